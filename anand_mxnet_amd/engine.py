"""Execution engine (parity with src/engine/ — ThreadedEnginePerDevice /
NaiveEngine and include/mxnet/engine.h).

MI355X-native design: the reference's engine exists to order CUDA work from
many frontend threads by variable dependencies.  On this stack, ordering is
the HIP stream model itself — ops are stream-ordered by the HIP runtime, and
the engine's job reduces to (a) stream management per device (compute /
H2D / D2H copy / comm streams, so copies and RCCL collectives overlap
compute — reference: MXNET_GPU_COPY_NTHREADS workers + FnProperty lanes),
(b) sync points (WaitForVar == tensor-ready event, WaitForAll == device
sync), and (c) a serial debug oracle (NaiveEngine semantics:
MXA_ENGINE_TYPE=NaiveEngine syncs after every op for deterministic
debugging, reference env MXNET_ENGINE_TYPE).

The native C++ side (src/engine.cpp) owns the raw hipStream_t pool; Python
wraps them as torch ExternalStreams so torch ops run on engine streams.
"""
from __future__ import annotations

import os
import torch

__all__ = ["Engine", "get", "naive_mode", "wait_all", "copy_stream",
           "comm_stream", "compute_stream"]


class Engine:
    _inst = None

    def __init__(self):
        self.engine_type = os.environ.get("MXA_ENGINE_TYPE",
                                          os.environ.get("MXNET_ENGINE_TYPE",
                                                         "StreamEngine"))
        self._copy_streams = {}
        self._comm_streams = {}
        self._native = None
        if torch.cuda.is_available():
            from .base import lib
            self._native = lib()

    @property
    def is_naive(self) -> bool:
        return self.engine_type == "NaiveEngine"

    def _get_stream(self, cache, device):
        if device not in cache:
            if self._native is not None:
                handle = self._native.engine_new_stream(device)
                cache[device] = torch.cuda.ExternalStream(handle, device=device)
            else:
                cache[device] = torch.cuda.Stream(device=device)
        return cache[device]

    def copy_stream(self, device: int):
        """Dedicated H2D/D2H stream (reference: kCopyFromGPU/kCopyToGPU lanes)."""
        return self._get_stream(self._copy_streams, device)

    def comm_stream(self, device: int):
        """Dedicated RCCL stream so collectives overlap backward compute."""
        return self._get_stream(self._comm_streams, device)

    def new_event(self):
        """VarHandle-style sync primitive (reference Engine::NewVariable/
        WaitForVar): record on a producing stream, wait from a consumer."""
        if self._native is not None:
            return self._native.engine_new_event()
        return None

    def record_event(self, ev, stream=None):
        if self._native is not None and ev is not None:
            s = stream.cuda_stream if stream is not None else                 torch.cuda.current_stream().cuda_stream
            self._native.engine_record_event(ev, s)

    def stream_wait_event(self, stream, ev):
        if self._native is not None and ev is not None:
            self._native.engine_stream_wait_event(stream.cuda_stream, ev)

    def wait_for_var(self, ev):
        """Block the host until the producer stream passed the event."""
        if self._native is not None and ev is not None:
            self._native.engine_sync_event(ev)

    def wait_all(self):
        if torch.cuda.is_available():
            for i in range(torch.cuda.device_count()):
                torch.cuda.synchronize(i)

    def after_op(self):
        """NaiveEngine oracle: serialize after every op."""
        if self.is_naive and torch.cuda.is_available():
            torch.cuda.synchronize()


class DepEngine:
    """The native variable-dependency scheduler (src/kernels/engine.hip —
    reference parity: ThreadedEngine var queues, ThreadedEnginePerDevice
    workers, NaiveEngine oracle via MXA_ENGINE_TYPE=NaiveEngine).

    Operations are pushed with explicit read/write variable lists and run
    on worker threads (per-device HIP streams for GPU payloads); the
    public sync points are wait_for_var / wait_all, and exceptions from
    payloads resurface there (reference exception contract)."""

    _inst = None

    def __init__(self):
        from .base import lib
        self._c = lib()
        if self._c is None or not hasattr(self._c, "dep_new_var"):
            raise RuntimeError("native engine not built")

    @classmethod
    def get(cls) -> "DepEngine":
        if cls._inst is None:
            cls._inst = DepEngine()
        return cls._inst

    @classmethod
    def available(cls) -> bool:
        try:
            cls.get()
            return True
        except Exception:
            return False

    @property
    def engine_type(self) -> str:
        return self._c.dep_engine_type()

    def new_var(self) -> int:
        return self._c.dep_new_var()

    def delete_var(self, var: int):
        self._c.dep_delete_var(var)

    def var_version(self, var: int) -> int:
        return self._c.dep_var_version(var)

    def push(self, fn, reads=(), writes=(), device=-1):
        self._c.dep_push(fn, list(reads), list(writes), device)

    def push_allreduce(self, tensors, average, reads, writes, device=-1):
        """Native no-GIL payload: in-place sum/average + broadcast."""
        self._c.dep_push_allreduce(tensors, average, list(reads),
                                   list(writes), device)

    def wait_for_var(self, var: int):
        self._c.dep_wait_for_var(var)

    def wait_all(self):
        self._c.dep_wait_all()

    def var_of(self, tensor) -> int:
        """Lazy engine variable attached to a tensor's storage."""
        v = getattr(tensor, "_mxa_dep_var", None)
        if v is None:
            v = self.new_var()
            tensor._mxa_dep_var = v
        return v


def get() -> Engine:
    if Engine._inst is None:
        Engine._inst = Engine()
    return Engine._inst


def naive_mode() -> bool:
    return get().is_naive


def wait_all():
    get().wait_all()


def copy_stream(device=0):
    return get().copy_stream(device)


def comm_stream(device=0):
    return get().comm_stream(device)


def compute_stream(device=0):
    return torch.cuda.current_stream(device)
