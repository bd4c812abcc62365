"""Runtime kernel compilation (parity with python/mxnet/rtc.py —
mx.rtc.CudaModule over NVRTC; here HipModule over hipRTC).

Compile HIP C++ source at runtime for gfx950 and launch kernels on
NDArrays, stream-ordered on the current torch/HIP stream.

    mod = mx.rtc.HipModule(r'''
        extern "C" __global__ void axpy(const float* x, float* y, float a,
                                        int n) {
            int i = blockIdx.x * blockDim.x + threadIdx.x;
            if (i < n) y[i] += a * x[i];
        }''')
    k = mod.get_kernel("axpy", "const float*, float*, float, int")
    k.launch((x, y, 3.0, 1000), mx.gpu(0), (250, 1, 1), (4, 1, 1))
"""
from __future__ import annotations

import ctypes
import ctypes.util
import re

import torch

from .base import MXNetError
from .ndarray.ndarray import NDArray

__all__ = ["HipModule", "CudaModule"]

_hiprtc = None
_hip = None


def _load_libs():
    global _hiprtc, _hip
    if _hiprtc is None:
        _hiprtc = ctypes.CDLL("libhiprtc.so")
    if _hip is None:
        _hip = ctypes.CDLL("libamdhip64.so")
    return _hiprtc, _hip


def _check_rtc(ret, prog=None):
    if ret != 0:
        msg = f"hiprtc error {ret}"
        if prog is not None:
            sz = ctypes.c_size_t()
            _hiprtc.hiprtcGetProgramLogSize(prog, ctypes.byref(sz))
            buf = ctypes.create_string_buffer(sz.value + 1)
            _hiprtc.hiprtcGetProgramLog(prog, buf)
            msg += ": " + buf.value.decode(errors="replace")
        raise MXNetError(msg)


def _check_hip(ret):
    if ret != 0:
        raise MXNetError(f"hip error {ret}")


class HipModule:
    """Compile HIP source with hipRTC; get_kernel returns launchable
    kernels (reference CudaModule.get_kernel signature)."""

    def __init__(self, source, options=(), exports=()):
        rtc, hip = _load_libs()
        prog = ctypes.c_void_p()
        _check_rtc(rtc.hiprtcCreateProgram(
            ctypes.byref(prog), source.encode(), b"mxa_rtc.cu", 0, None,
            None))
        opts = [b"--offload-arch=gfx950", b"-O3", b"-std=c++17"] + \
            [o.encode() for o in options]
        arr = (ctypes.c_char_p * len(opts))(*opts)
        ret = rtc.hiprtcCompileProgram(prog, len(opts), arr)
        _check_rtc(ret, prog)
        sz = ctypes.c_size_t()
        _check_rtc(rtc.hiprtcGetCodeSize(prog, ctypes.byref(sz)))
        code = ctypes.create_string_buffer(sz.value)
        _check_rtc(rtc.hiprtcGetCode(prog, code))
        rtc.hiprtcDestroyProgram(ctypes.byref(prog))
        self._code = code
        self._modules = {}  # device -> hipModule

    def _module_for(self, dev_id):
        m = self._modules.get(dev_id)
        if m is None:
            _, hip = _load_libs()
            _check_hip(hip.hipSetDevice(dev_id))
            mod = ctypes.c_void_p()
            _check_hip(hip.hipModuleLoadData(ctypes.byref(mod), self._code))
            self._modules[dev_id] = mod
            m = mod
        return m

    def get_kernel(self, name, signature):
        """signature: comma-separated C types (reference parity), e.g.
        'const float*, float*, float, int'."""
        types = [tt.strip() for tt in signature.split(",") if tt.strip()]
        return HipKernel(self, name, types)


CudaModule = HipModule  # reference-API alias


_SCALAR_CTYPES = {
    "float": ctypes.c_float, "double": ctypes.c_double,
    "int": ctypes.c_int, "unsigned": ctypes.c_uint,
    "unsigned int": ctypes.c_uint, "long": ctypes.c_long,
    "long long": ctypes.c_longlong, "size_t": ctypes.c_size_t,
    "int64_t": ctypes.c_int64, "uint64_t": ctypes.c_uint64,
    "int32_t": ctypes.c_int32, "uint32_t": ctypes.c_uint32,
}


class HipKernel:
    def __init__(self, module, name, types):
        self._module = module
        self._name = name.encode()
        self._types = types

    def launch(self, args, ctx, grid_dims, block_dims, shared_mem=0):
        _, hip = _load_libs()
        dev = ctx.device_id if hasattr(ctx, "device_id") else int(ctx)
        mod = self._module._module_for(dev)
        fn = ctypes.c_void_p()
        _check_hip(hip.hipModuleGetFunction(ctypes.byref(fn), mod,
                                            self._name))
        cargs = []
        for a, tdesc in zip(args, self._types):
            if isinstance(a, NDArray):
                cargs.append(ctypes.c_void_p(a._t.data_ptr()))
            elif isinstance(a, torch.Tensor):
                cargs.append(ctypes.c_void_p(a.data_ptr()))
            else:
                base = re.sub(r"\bconst\b|\s+\*|\*", "", tdesc).strip()
                ct = _SCALAR_CTYPES.get(base, ctypes.c_float
                                        if isinstance(a, float)
                                        else ctypes.c_int)
                cargs.append(ct(a))
        ptrs = (ctypes.c_void_p * len(cargs))(
            *[ctypes.cast(ctypes.byref(c), ctypes.c_void_p) for c in cargs])
        stream = ctypes.c_void_p(torch.cuda.current_stream(dev).cuda_stream)
        _check_hip(hip.hipModuleLaunchKernel(
            fn, grid_dims[0], grid_dims[1], grid_dims[2],
            block_dims[0], block_dims[1], block_dims[2],
            shared_mem, stream, ptrs, None))
