"""Operator library.  Importing this package registers all ops (reference
parity: the NNVM registry built by src/operator/*'s static registrations)."""
from . import registry
from . import core  # noqa: F401 - registers core ops
from . import nn    # noqa: F401 - registers nn ops
from . import optim  # noqa: F401 - registers optimizer ops

# GPU (HIP kernel) implementations register themselves if the native
# extension is importable; on CPU-only boxes this is a silent no-op.
try:
    from . import hip  # noqa: F401
except Exception:  # pragma: no cover
    pass

from .registry import OPS, RAW, GPU_IMPL, register, register_gpu, gpu_impl, \
    has_gpu_impl, get  # noqa: F401
