"""In-tree build of the MI355X-native extension (_C).

  PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

hipcc cross-compiles gfx950 without a GPU; the built .so lives inside the
package (anand_mxnet_amd/_C*.so) so gpurun snapshots ship it.
"""
import os
import glob

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

ROOT = os.path.dirname(os.path.abspath(__file__))
SRC = os.path.join(ROOT, "anand_mxnet_amd", "src")

sources = [os.path.join(SRC, "bindings.hip")] + \
    sorted(glob.glob(os.path.join(SRC, "kernels", "*.hip")))

setup(
    name="anand_mxnet_amd_native",
    ext_modules=[
        CUDAExtension(
            name="anand_mxnet_amd._C",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
