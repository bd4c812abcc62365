"""GPU dispatch: hand-written CDNA4 kernels wrapped as torch.autograd
Functions and registered into the op registry (ops.registry.GPU_IMPL).

Contract: these run for bf16 tensors on gpu(i).  fp32 GPU tensors use the
torch-ROCm substrate (documented in docs/PARITY.md); bf16 is the
performance dtype of every target config.
"""
from __future__ import annotations

import torch

from ..registry import register_gpu
from ...base import lib

_C = lib()
if _C is None:  # pragma: no cover
    raise ImportError("native extension not built")

_seed_counter = [12345]


def _next_seed():
    _seed_counter[0] = (_seed_counter[0] * 6364136223846793005 + 1442695040888963407) % (1 << 63)
    return _seed_counter[0]


def reset_seed(n):
    """Reseed the native RNG salt stream (dropout masks, flash dropout
    replay, sampling) — called by mx.random.seed so runs are
    reproducible under seeding (reference: mx.random.seed seeds the
    engine's kRandom/kParallelRandom resources)."""
    _seed_counter[0] = int(n) % (1 << 63)


# ---------------------------------------------------------------------------
# GEMM / linear
# ---------------------------------------------------------------------------

class _GemmTN(torch.autograd.Function):
    """C = A @ B^T over the MFMA TN kernel; A [.., M, K], B [.., N, K]."""

    @staticmethod
    def forward(ctx, a, b):
        ctx.save_for_backward(a, b)
        return _C.gemm_tn(a.contiguous(), b.contiguous(), None, False)

    @staticmethod
    def backward(ctx, dy):
        a, b = ctx.saved_tensors
        dy = dy.contiguous()
        N = b.shape[-2]
        M = a.shape[-2]
        da = db = None
        if ctx.needs_input_grad[0]:
            # dA[M,K] = dY[M,N] @ B[N,K] — the NN-form kernel consumes
            # dY (N-contiguous) and B (K-contiguous) in place, no B^T copy
            if a.dim() == 2 and N % 64 == 0 and a.shape[-1] % 8 == 0:
                da = _C.gemm_nn(dy, b)
            elif N % 64 == 0:
                da = _C.gemm_tn(dy, b.transpose(-1, -2).contiguous(), None,
                                False)
            else:  # non-tileable reduce dim: library GEMM
                da = torch.matmul(dy, b)
        if ctx.needs_input_grad[1]:
            if a.dim() == 2:
                db = _wg_or_matmul(dy, a, a.dtype)  # [N,K]
            elif M % 64 == 0:
                db = _C.gemm_tn(dy.transpose(-1, -2).contiguous(),
                                a.transpose(-1, -2).contiguous(), None,
                                False)
            else:
                db = torch.matmul(dy.transpose(-1, -2), a)
        return da, db



def _tn_or_matmul(a, b):
    """TN MFMA kernel when the reduce dim tiles, else library GEMM."""
    if a.size(-1) % 64 == 0:
        return _GemmTN.apply(a.contiguous(), b.contiguous())
    return torch.matmul(a, b.transpose(-1, -2))


def _wg_or_matmul(dy, x, out_dtype):
    """WG MFMA kernel for dW = dy^T x.

    The WG kernel emits bf16 directly (split slots folded + cast in its
    epilogue), so .to() is a no-op for the bf16 weights of every config.
    Narrow heads (e.g. the BERT NSP 2-class dy) are zero-padded to the
    %8 output quantum and sliced back — a row prefix stays contiguous —
    instead of falling back to a library GEMM."""
    if x.size(-1) % 8 != 0:
        return torch.matmul(dy.t(), x).to(out_dtype)
    n = dy.size(-1)
    if n % 8 != 0:
        dyp = torch.nn.functional.pad(dy, (0, (-n) % 8))
        return _C.gemm_wg(dyp.contiguous(), x)[:n].to(out_dtype)
    return _C.gemm_wg(dy, x).to(out_dtype)


@register_gpu("gemm")
def gemm(a, b, transpose_a, transpose_b):
    A = a.transpose(-1, -2) if transpose_a else a
    Bt = b if transpose_b else b.transpose(-1, -2)
    if A.size(-1) % 64 != 0:
        # reduce-dim not tileable: fall back to library GEMM (plain GEMMs
        # may use rocBLAS; the tiled hot paths are all %64)
        return torch.matmul(A, Bt.transpose(-1, -2))
    return _GemmTN.apply(A.contiguous(), Bt.contiguous())


@register_gpu("batch_gemm")
def batch_gemm(a, b, transpose_a, transpose_b):
    A = a.transpose(-1, -2) if transpose_a else a
    Bt = b if transpose_b else b.transpose(-1, -2)
    if A.size(-1) % 64 != 0:
        return torch.matmul(A, Bt.transpose(-1, -2))
    return _GemmTN.apply(A.contiguous(), Bt.contiguous())


class _Linear(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, bias):
        ctx.save_for_backward(x, w)
        ctx.has_bias = bias is not None
        return _C.gemm_tn(x.contiguous(), w.contiguous(),
                          bias.contiguous() if bias is not None else None,
                          False)

    @staticmethod
    def backward(ctx, dy):
        x, w = ctx.saved_tensors
        dy = dy.contiguous()
        dx = dw = db = None
        if ctx.needs_input_grad[0]:
            if w.shape[0] % 64 == 0 and w.shape[1] % 8 == 0:
                dx = _C.gemm_nn(dy, w)  # no W^T copy (NN-form kernel)
            elif w.shape[1] % 8 == 0:
                # non-%64 class dim (e.g. the 1000-way head): pad the
                # reduce dim to 64 — zero rows of W contribute nothing —
                # and stay on the NN MFMA kernel (no rocBLAS in the step)
                N = w.shape[0]
                npad = (-N) % 64
                dyp = torch.nn.functional.pad(dy, (0, npad))
                wp = torch.nn.functional.pad(w, (0, 0, 0, npad))
                dx = _C.gemm_nn(dyp, wp)
            else:
                dx = torch.matmul(dy, w)
        if ctx.needs_input_grad[1]:
            want_db = ctx.has_bias and ctx.needs_input_grad[2]
            if (want_db and x.size(-1) % 8 == 0 and dy.size(-1) % 8 == 0):
                # fused dW + db: the wg kernel folds colsum(dy) from the
                # A fragments it already holds (no extra dy read)
                dw, db = _C.gemm_wg_bias(dy, x)
                dw = dw.to(w.dtype)
            else:
                dw = _wg_or_matmul(dy, x, w.dtype)
                if want_db:
                    db = _C.colsum(dy)
        elif ctx.has_bias and ctx.needs_input_grad[2]:
            db = _C.colsum(dy)
        return dx, dw, db


@register_gpu("linear")
def linear(x, w, bias):
    if x.size(-1) % 64 != 0:
        return torch.nn.functional.linear(x, w, bias)
    x2 = x.reshape(-1, x.shape[-1])
    out = _Linear.apply(x2.contiguous(), w,
                        bias if bias is None else bias.to(x.dtype))
    return out.reshape(x.shape[:-1] + (w.shape[0],))


# ---------------------------------------------------------------------------
# Convolution (NHWC bf16)
# ---------------------------------------------------------------------------

class _Conv2dNHWC(torch.autograd.Function):
    """Implicit-GEMM conv; weight logical layout (K, R, S, C)."""

    @staticmethod
    def forward(ctx, x, w, bias, stride, pad, dil, stats_out=None):
        # stats_out: optional list; when the kernel route supports it, the
        # per-channel BN sum/sumsq accumulated in the GEMM epilogue is
        # appended so the caller can hand it to the following BatchNorm
        # (skips bn_stats_partial's re-read of y; see gemm_tn.hip).
        x = x.contiguous()
        K, R, S, C = w.shape
        ctx.save_for_backward(x, w)
        ctx.conf = (stride, pad, dil)
        ctx.has_bias = bias is not None
        if R == 1 and S == 1 and stride == (1, 1) and pad == (0, 0):
            # pointwise conv IS a plain GEMM: route through gemm_tn so
            # the 256-tile 8-phase kernel picks it up (the implicit-GEMM
            # gather loader adds address math for nothing here)
            N, H, W_, _ = x.shape
            b32 = bias.float() if bias is not None else None
            if stats_out is not None:
                res = _C.gemm_tn_stats(x.reshape(N * H * W_, C),
                                       w.reshape(K, C), b32, False)
                if len(res) == 2:
                    stats_out.append(res[1])
                return res[0].reshape(N, H, W_, K)
            y2 = _C.gemm_tn(x.reshape(N * H * W_, C), w.reshape(K, C),
                            b32, False)
            return y2.reshape(N, H, W_, K)
        w_rskc = w.permute(1, 2, 0, 3).contiguous()
        b32 = bias.float() if bias is not None else None
        if stats_out is not None:
            res = _C.conv2d_fwd_stats(x, w_rskc, b32, list(stride),
                                      list(pad), list(dil), False)
            if len(res) == 2:
                stats_out.append(res[1])
            return res[0]
        y = _C.conv2d_fwd(x, w_rskc, b32,
                          list(stride), list(pad), list(dil), False)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w = ctx.saved_tensors
        stride, pad, dil = ctx.conf
        dy = dy.contiguous()
        K, R, S, C = w.shape
        dx = dw = db = None
        if ctx.needs_input_grad[0]:
            if R == 1 and S == 1 and stride == (1, 1) and pad == (0, 0) \
                    and K % 64 == 0 and C % 8 == 0:
                # pointwise: dx = dy @ W, the NN-form kernel (no W^T copy)
                N, P, Q, _ = dy.shape
                dx = _C.gemm_nn(dy.reshape(N * P * Q, K),
                                w.reshape(K, C)).reshape(x.shape)
            elif R == 3 and S == 3 and stride == (2, 2) and pad == (1, 1) \
                    and dil == (1, 1) and x.shape[1] % 2 == 0 \
                    and x.shape[2] % 2 == 0 and K % 64 == 0 and C % 8 == 0:
                # stride-2 3x3 (ResNet transition): the scatter-inverse
                # gather GEMM wastes 3/4 of its rows on parity predicates.
                # Class-decompose by (h%2, w%2): each input-pixel class
                # uses a fixed (r, s) weight subset, so the dgrad becomes
                # 9 DENSE [NPQ,K]x[K,C] NN GEMMs (total FLOPs = fwd) plus
                # shifted quarter-grid adds.
                N, P, Q, _ = dy.shape
                dy2 = dy.reshape(N * P * Q, K)

                def g(r, s_):
                    return _C.gemm_nn(dy2, w[:, r, s_, :].contiguous()) \
                        .reshape(N, P, Q, C)

                dx = torch.zeros(x.shape, dtype=dy.dtype, device=dy.device)
                # h even, w even: r=1, s=1, (p,q)=(hh,ww)
                dx[:, 0::2, 0::2, :] = g(1, 1)
                # h even, w odd: s=0 -> q=ww+1 ; s=2 -> q=ww
                c01 = dx[:, 0::2, 1::2, :]
                c01.copy_(g(1, 2))
                c01[:, :, :Q - 1, :] += g(1, 0)[:, :, 1:, :]
                # h odd, w even: r=0 -> p=hh+1 ; r=2 -> p=hh
                c10 = dx[:, 1::2, 0::2, :]
                c10.copy_(g(2, 1))
                c10[:, :P - 1, :, :] += g(0, 1)[:, 1:, :, :]
                # h odd, w odd: four corner taps
                c11 = dx[:, 1::2, 1::2, :]
                c11.copy_(g(2, 2))
                c11[:, :, :Q - 1, :] += g(2, 0)[:, :, 1:, :]
                c11[:, :P - 1, :, :] += g(0, 2)[:, 1:, :, :]
                c11[:, :P - 1, :Q - 1, :] += g(0, 0)[:, 1:, 1:, :]
                dx = dx.contiguous()
            elif R == 1 and S == 1 and (stride[0] > 1 or stride[1] > 1) \
                    and pad == (0, 0):
                # strided 1x1 (ResNet downsample): only every (sh, sw)-th
                # input pixel has gradient — dense GEMM over the P*Q
                # subgrid + strided scatter instead of a scatter-inverse
                # gather GEMM where 3/4 of the predicate rows are zero.
                N, P, Q, _ = dy.shape
                dy2 = dy.reshape(N * P * Q, K)
                sub = _C.gemm_nn(dy2, w.reshape(K, C))  # [NPQ, C], no W^T
                dx = torch.zeros(x.shape, dtype=dy.dtype, device=dy.device)
                dx[:, ::stride[0], ::stride[1], :] = sub.reshape(N, P, Q, C)
            else:
                w_rsck = w.permute(1, 2, 3, 0).contiguous()
                if K % 64 != 0:
                    # dgrad reduces over K: zero-pad dy/w channels to the
                    # tile width (zeros contribute nothing)
                    Kp = (K + 63) // 64 * 64
                    dy_p = torch.nn.functional.pad(dy, (0, Kp - K)) \
                        .contiguous()
                    w_p = torch.nn.functional.pad(w_rsck, (0, Kp - K)) \
                        .contiguous()
                    dx = _C.conv2d_dgrad(dy_p, w_p, list(x.shape),
                                         list(stride), list(pad),
                                         list(dil))
                else:
                    dx = _C.conv2d_dgrad(dy, w_rsck, list(x.shape),
                                         list(stride), list(pad),
                                         list(dil))
        if ctx.needs_input_grad[1]:
            if R == 1 and S == 1 and stride == (1, 1) and pad == (0, 0) \
                    and K % 8 == 0 and C % 8 == 0:
                # pointwise wgrad: plain WG GEMM; the transposed fold
                # emits [K,C] bf16 == (K,1,1,C) contiguous directly
                N, P, Q, _ = dy.shape
                dw = _C.gemm_wg(x.reshape(N * P * Q, C),
                                dy.reshape(N * P * Q, K),
                                True).reshape(K, 1, 1, C)
            else:
                # fold epilogue returns (K,R,S,C) bf16 — no permute/cast
                dw = _C.conv2d_wgrad(x, dy, [R, S], list(stride),
                                     list(pad), list(dil))
        if ctx.has_bias and ctx.needs_input_grad[2]:
            db = _C.colsum(dy)
        return dx, dw, db, None, None, None, None


class _Conv2dIm2col(torch.autograd.Function):
    """Small-C path (the 7x7 C=3 stem): materialized im2col + TN/WG GEMM."""

    @staticmethod
    def forward(ctx, x, w, bias, stride, pad, dil):
        K, R, S, C = w.shape
        N, H, W_, _ = x.shape
        sh, sw = stride
        ph, pw = pad
        P = (H + 2 * ph - (R - 1) * dil[0] - 1) // sh + 1
        Q = (W_ + 2 * pw - (S - 1) * dil[1] - 1) // sw + 1
        kdim = R * S * C
        kpadded = ((kdim + 63) // 64) * 64
        # dedicated gather kernel: writes the padded GEMM operand directly
        # (replaces torch pad + advanced-indexing + pad + contiguous chain)
        A = _C.im2col_pad(x.contiguous(), [R, S], list(stride), list(pad),
                          list(dil), kpadded)
        kpad = kpadded - kdim
        B = w.reshape(K, kdim)
        if kpad:
            B = torch.nn.functional.pad(B, (0, kpad))
        ctx.save_for_backward(A, w)
        ctx.dims = (N, P, Q, K, kdim)
        ctx.xshape = tuple(x.shape)
        ctx.rs = (R, S)
        ctx.conf = (stride, pad, dil)
        ctx.has_bias = bias is not None
        y = _C.gemm_tn(A, B.contiguous(),
                       bias.float() if bias is not None else None, False)
        return y.reshape(N, P, Q, K)

    @staticmethod
    def backward(ctx, dy):
        A, w = ctx.saved_tensors
        N, P, Q, K, kdim = ctx.dims
        dy2 = dy.reshape(N * P * Q, K).contiguous()
        dw = db = None
        if ctx.needs_input_grad[1]:
            if ctx.has_bias and ctx.needs_input_grad[2]:
                dwf, db = _C.gemm_wg_bias(dy2, A)  # fused colsum(dy)
            else:
                dwf = _C.gemm_wg(dy2, A)   # [K, kdim_padded] bf16
            dw = dwf[:, :kdim].reshape(w.shape).to(w.dtype)
        elif ctx.has_bias and ctx.needs_input_grad[2]:
            db = _C.colsum(dy2)
        dx = None
        if ctx.needs_input_grad[0]:
            # col2im dgrad (inner small-C convs, e.g. Inception branches):
            # dcols = dy @ W, scatter-add via F.fold.  Our k-order is
            # (r, s, c); fold expects (c, r, s).
            stride, pad, dil = ctx.conf
            N2, H2, W2, C2 = ctx.xshape
            R2, S2 = ctx.rs
            dcols = torch.matmul(dy2.float(),
                                 w.reshape(K, kdim).float())
            dcols = dcols.reshape(N2, P * Q, R2, S2, C2) \
                .permute(0, 4, 2, 3, 1).reshape(N2, C2 * R2 * S2, P * Q)
            dxn = torch.nn.functional.fold(
                dcols, (H2, W2), (R2, S2), dilation=tuple(dil),
                padding=tuple(pad), stride=tuple(stride))  # [N, C, H, W]
            dx = dxn.permute(0, 2, 3, 1).contiguous().to(dy.dtype)
        return dx, dw, db, None, None, None


class _DwConvNHWC(torch.autograd.Function):
    """Depthwise conv (groups == C), weight (C, R, S, 1) -> [R,S,C]."""

    @staticmethod
    def forward(ctx, x, w, bias, stride, pad, dil):
        x = x.contiguous()
        w_rsc = w[:, :, :, 0].permute(1, 2, 0).contiguous()  # [R,S,C]
        ctx.save_for_backward(x, w_rsc)
        ctx.conf = (stride, pad, dil, w.shape)
        ctx.has_bias = bias is not None
        return _C.dwconv_fwd(x, w_rsc,
                             bias.float() if bias is not None else None,
                             list(stride), list(pad), list(dil))

    @staticmethod
    def backward(ctx, dy):
        x, w_rsc = ctx.saved_tensors
        stride, pad, dil, wshape = ctx.conf
        dy = dy.contiguous()
        dx = dw = db = None
        if ctx.needs_input_grad[0]:
            dx = _C.dwconv_dgrad(dy, w_rsc, list(x.shape), list(stride),
                                 list(pad))
        if ctx.needs_input_grad[1]:
            dwf = _C.dwconv_wgrad(x, dy, [w_rsc.shape[0], w_rsc.shape[1]],
                                  list(stride), list(pad))  # [R,S,C] f32
            dw = dwf.permute(2, 0, 1).unsqueeze(-1).to(x.dtype)                 .contiguous()
        if ctx.has_bias and ctx.needs_input_grad[2]:
            db = _C.colsum(dy)
        return dx, dw, db, None, None, None


import os

# conv->BN fused statistics: measured NEGATIVE on MI355X (profiles/
# RESULTS.md): the cross-block per-column atomic reduction in the GEMM
# epilogue costs more than the near-roofline bn_stats_partial re-read it
# saves (7142 vs 7328 samples/s on the b256 flagship even with 32-way
# slotted accumulators).  Machinery kept correct + tested; off by default.
_CONV_BN_STATS = os.environ.get("MXA_CONV_BN_STATS", "0") == "1"


@register_gpu("conv2d_nhwc")
def conv2d_nhwc(x, w, bias, stride, pad, dil, groups, collect_stats=False,
                relu=False):
    if relu:
        # fused conv+relu epilogue: inference-only (conv2d_fwd's RELU
        # template; backward has no mask).  Under autograd fall back to
        # a separate relu so grads stay correct.
        if torch.is_grad_enabled() and (x.requires_grad or w.requires_grad):
            return torch.relu(conv2d_nhwc(x, w, bias, stride, pad, dil,
                                          groups, collect_stats))
        if groups == 1 and w.shape[3] % 64 == 0:
            stride, pad, dil = tuple(stride), tuple(pad), tuple(dil)
            x = x.contiguous()
            K, R, S, C = w.shape
            b32 = bias.float() if bias is not None else None
            if R == 1 and S == 1 and stride == (1, 1) and pad == (0, 0):
                N, H, W_, _ = x.shape
                y2 = _C.gemm_tn(x.reshape(N * H * W_, C),
                                w.reshape(K, C), b32, True)
                return y2.reshape(N, H, W_, K)
            w_rskc = w.permute(1, 2, 0, 3).contiguous()
            return _C.conv2d_fwd(x, w_rskc, b32, list(stride), list(pad),
                                 list(dil), True)
        return torch.relu(conv2d_nhwc(x, w, bias, stride, pad, dil,
                                      groups, collect_stats))
    if groups != 1:
        C_in = x.shape[3]
        if groups == C_in and w.shape[0] == C_in and w.shape[3] == 1                 and C_in % 8 == 0 and tuple(dil) == (1, 1):
            return _DwConvNHWC.apply(x, w, bias, tuple(stride),
                                     tuple(pad), tuple(dil))
        # general grouped conv (ResNeXt-style): per-group channel slices
        # through the dense implicit-GEMM kernels.  Each group is a full
        # conv of (K/G, R, S, C/G); channel-contiguous NHWC slicing makes
        # the per-group inputs cheap views + one concat at the end.
        K = w.shape[0]
        if K % groups or C_in % groups:
            raise NotImplementedError(
                "grouped conv: channels must divide groups")
        kg = K // groups
        outs = []
        for g in range(groups):
            xg = x[..., g * (C_in // groups):(g + 1) * (C_in // groups)] \
                .contiguous()
            wg = w[g * kg:(g + 1) * kg]
            bg = bias[g * kg:(g + 1) * kg] if bias is not None else None
            outs.append(conv2d_nhwc(xg, wg, bg, stride, pad, dil, 1))
        return torch.cat(outs, dim=3)
    stride, pad, dil = tuple(stride), tuple(pad), tuple(dil)
    if w.shape[3] % 64 == 0:
        holder = [] if (collect_stats and _CONV_BN_STATS
                        and torch.is_grad_enabled()) else None
        y = _Conv2dNHWC.apply(x, w, bias, stride, pad, dil, holder)
        if holder:
            y._mxa_bn_stats = holder[0]
        return y
    return _Conv2dIm2col.apply(x, w, bias, stride, pad, dil)


# ---------------------------------------------------------------------------
# BatchNorm (NHWC)
# ---------------------------------------------------------------------------

class _BatchNormNHWC(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, resid, gamma, beta, rmean, rvar, eps, momentum,
                training, fuse_relu, psums=None):
        x = x.contiguous()
        g32 = gamma.float().contiguous()
        b32 = beta.float().contiguous()
        fuse_add = resid is not None
        if training:
            y, save_mean, save_rstd = _C.bn_fwd_train(
                x, g32, b32, rmean, rvar, eps, momentum, fuse_relu,
                resid.contiguous() if fuse_add else None, psums)
            if fuse_add:
                # junction form relu(bn(x)+r): mask comes from y > 0
                ctx.save_for_backward(x, g32, b32, save_mean, save_rstd, y)
            else:
                ctx.save_for_backward(x, g32, b32, save_mean, save_rstd)
            ctx.relu = fuse_relu
            ctx.fuse_add = fuse_add
            ctx.gdtype = gamma.dtype
            return y
        ctx.inference = True
        return _C.bn_fwd_infer(x, g32, b32, rmean.contiguous(),
                               rvar.contiguous(), eps, fuse_relu,
                               resid.contiguous() if fuse_add else None)

    @staticmethod
    def backward(ctx, dy):
        if getattr(ctx, "inference", False):
            raise RuntimeError("BatchNorm inference backward not supported")
        if ctx.fuse_add:
            x, g32, b32, save_mean, save_rstd, y = ctx.saved_tensors
            dr = _C.act_bwd(0, y, dy.contiguous())  # relu mask from y
            dx, dgamma, dbeta = _C.bn_bwd(x, dr, g32, b32,
                                          save_mean, save_rstd, False)
            return (dx, dr, dgamma.to(ctx.gdtype), dbeta.to(ctx.gdtype),
                    None, None, None, None, None, None, None)
        x, g32, b32, save_mean, save_rstd = ctx.saved_tensors
        dx, dgamma, dbeta = _C.bn_bwd(x, dy.contiguous(), g32, b32,
                                      save_mean, save_rstd, ctx.relu)
        return (dx, None, dgamma.to(ctx.gdtype), dbeta.to(ctx.gdtype), None,
                None, None, None, None, None, None)


@register_gpu("batchnorm_nhwc")
def batchnorm_nhwc(x, gamma, beta, rmean, rvar, eps, momentum, training,
                   fuse_relu, addend=None):
    if x.dtype != torch.bfloat16 or x.shape[-1] % 8 != 0:
        # torch path for non-native dtype/shape
        out = torch.nn.functional.batch_norm(
            x.permute(0, 3, 1, 2), rmean, rvar, gamma, beta, training,
            1.0 - momentum if training else 0.0, eps)
        if addend is not None:
            out = out + addend.permute(0, 3, 1, 2)
            out = torch.relu(out)
        elif fuse_relu:
            out = torch.relu(out)
        return out.permute(0, 2, 3, 1).contiguous()
    # fused conv->BN statistics: the producing conv left its per-channel
    # sum/sumsq on the tensor (conv2d_nhwc collect_stats); x.contiguous()
    # below is a no-op for the conv output, so the attribute's tensor
    # identity is the one the stats describe.
    psums = getattr(x, "_mxa_bn_stats", None) if training else None
    if psums is not None and psums.numel() != 2 * x.shape[-1]:
        psums = None
    return _BatchNormNHWC.apply(x, addend, gamma, beta, rmean, rvar, eps,
                                momentum, training, fuse_relu, psums)


# ---------------------------------------------------------------------------
# Pooling (NHWC)
# ---------------------------------------------------------------------------

class _MaxPoolNHWC(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, kernel, stride, pad):
        x = x.contiguous()
        y, arg = _C.maxpool(x, list(kernel), list(stride), list(pad))
        ctx.save_for_backward(arg)
        ctx.conf = (tuple(x.shape), kernel, stride, pad)
        return y

    @staticmethod
    def backward(ctx, dy):
        (arg,) = ctx.saved_tensors
        xshape, kernel, stride, pad = ctx.conf
        dx = _C.maxpool_bwd(dy.contiguous(), arg, list(xshape),
                            list(kernel), list(stride), list(pad))
        return dx, None, None, None


class _GlobalAvgPoolNHWC(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        x = x.contiguous()
        ctx.xshape = tuple(x.shape)
        return _C.global_avgpool(x)

    @staticmethod
    def backward(ctx, dy):
        return _C.global_avgpool_bwd(dy.contiguous(), list(ctx.xshape))


@register_gpu("pool_nhwc")
def pool_nhwc(x, kernel, pool_type, global_pool, stride, pad,
              count_include_pad):
    if x.dtype != torch.bfloat16 or x.shape[-1] % 8 != 0:
        import torch.nn.functional as F
        xc = x.permute(0, 3, 1, 2)
        if global_pool:
            out = (F.adaptive_max_pool2d(xc, 1) if pool_type == "max"
                   else F.adaptive_avg_pool2d(xc, 1))
        elif pool_type == "max":
            out = F.max_pool2d(xc, tuple(kernel), tuple(stride or kernel),
                               tuple(pad))
        else:
            out = F.avg_pool2d(xc, tuple(kernel), tuple(stride or kernel),
                               tuple(pad),
                               count_include_pad=count_include_pad)
        return out.permute(0, 2, 3, 1).contiguous()
    if global_pool:
        if pool_type == "avg":
            return _GlobalAvgPoolNHWC.apply(x)
        kernel = (x.shape[1], x.shape[2])
        stride, pad = (1, 1), (0, 0)
    stride = tuple(stride) if stride is not None else tuple(kernel)
    if pool_type == "max":
        return _MaxPoolNHWC.apply(x, tuple(kernel), stride, tuple(pad))
    # average pooling: no stored state needed; use autograd-free fwd +
    # custom bwd via the gap/avg kernels
    class _Avg(torch.autograd.Function):
        @staticmethod
        def forward(ctx2, xx):
            ctx2.conf = (tuple(xx.shape), tuple(kernel), stride, tuple(pad),
                         count_include_pad)
            return _C.avgpool(xx.contiguous(), list(kernel), list(stride),
                              list(pad), count_include_pad)

        @staticmethod
        def backward(ctx2, dy):
            import torch.nn.functional as F
            xshape, kern, strd, pd, cip = ctx2.conf
            # exact bwd via torch avg_pool grad on permuted layout
            dxc = torch.ops.aten.avg_pool2d_backward(
                dy.permute(0, 3, 1, 2), torch.empty(
                    (xshape[0], xshape[3], xshape[1], xshape[2]),
                    device=dy.device, dtype=dy.dtype), list(kern),
                list(strd), list(pd), False, cip, None)
            return dxc.permute(0, 2, 3, 1).contiguous()
    return _Avg.apply(x)


# ---------------------------------------------------------------------------
# softmax / layernorm / dropout / embedding / activations
# ---------------------------------------------------------------------------

class _Softmax(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, log_):
        y = _C.softmax_fwd(x.contiguous(), log_, 1.0)
        ctx.save_for_backward(y)
        ctx.log = log_
        return y

    @staticmethod
    def backward(ctx, dy):
        (y,) = ctx.saved_tensors
        return _C.softmax_bwd(y, dy.contiguous(), ctx.log, 1.0), None


@register_gpu("softmax")
def softmax(x):
    if x.dtype != torch.bfloat16 or x.shape[-1] % 8 != 0:
        return torch.nn.functional.softmax(x.float(), dim=-1).to(x.dtype)
    return _Softmax.apply(x, False)


@register_gpu("log_softmax")
def log_softmax(x):
    if x.dtype != torch.bfloat16 or x.shape[-1] % 8 != 0:
        return torch.nn.functional.log_softmax(x.float(), dim=-1).to(x.dtype)
    return _Softmax.apply(x, True)


class _LayerNorm(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, gamma, beta, eps):
        x = x.contiguous()
        g32 = gamma.float().contiguous()
        y, mean, rstd = _C.layernorm_fwd(x, g32, beta.float().contiguous(),
                                         eps)
        ctx.save_for_backward(x, g32, mean, rstd)
        ctx.gdtype = gamma.dtype
        return y

    @staticmethod
    def backward(ctx, dy):
        x, g32, mean, rstd = ctx.saved_tensors
        dx, dgamma, dbeta = _C.layernorm_bwd(x, dy.contiguous(), g32, mean,
                                             rstd)
        return dx, dgamma.to(ctx.gdtype), dbeta.to(ctx.gdtype), None


@register_gpu("layernorm")
def layernorm(x, gamma, beta, eps):
    if x.dtype != torch.bfloat16 or x.shape[-1] % 8 != 0:
        return torch.nn.functional.layer_norm(x, (x.shape[-1],), gamma,
                                              beta, eps)
    return _LayerNorm.apply(x, gamma, beta, eps)


class _Dropout(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, p):
        y, mask = _C.dropout_fwd(x.contiguous(), p, _next_seed())
        ctx.save_for_backward(mask)
        ctx.p = p
        return y

    @staticmethod
    def backward(ctx, dy):
        (mask,) = ctx.saved_tensors
        return _C.dropout_bwd(dy.contiguous(), mask, ctx.p), None


@register_gpu("dropout")
def dropout(x, p):
    if x.dtype != torch.bfloat16 or x.numel() % 8 != 0:
        return torch.nn.functional.dropout(x, p, training=True)
    return _Dropout.apply(x, p)


class _Embedding(torch.autograd.Function):
    @staticmethod
    def forward(ctx, idx, W):
        ctx.save_for_backward(idx)
        ctx.vocab = W.shape[0]
        ctx.wdtype = W.dtype
        return _C.embedding_fwd(idx, W.contiguous())

    @staticmethod
    def backward(ctx, dy):
        (idx,) = ctx.saved_tensors
        dW = _C.embedding_bwd(idx, dy.contiguous(), ctx.vocab)
        return None, dW.to(ctx.wdtype)


@register_gpu("embedding")
def embedding(idx, W):
    if W.dtype != torch.bfloat16 or W.shape[1] % 8 != 0:
        return torch.nn.functional.embedding(idx.long(), W)
    return _Embedding.apply(idx, W)


_ACT_IDS = {"relu": 0, "sigmoid": 1, "tanh": 2, "gelu": 3, "silu": 4,
            "gelu_tanh": 5}


class _Act(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, op):
        x = x.contiguous()
        ctx.save_for_backward(x)
        ctx.op = op
        return _C.act_fwd(op, x)

    @staticmethod
    def backward(ctx, dy):
        (x,) = ctx.saved_tensors
        return _C.act_bwd(ctx.op, x, dy.contiguous()), None


def _make_act(name, opid):
    def fn(x):
        if x.dtype != torch.bfloat16 or x.numel() % 8 != 0:
            return {"relu": torch.relu, "sigmoid": torch.sigmoid,
                    "tanh": torch.tanh,
                    "gelu": torch.nn.functional.gelu,
                    "silu": torch.nn.functional.silu,
                    "gelu_tanh": lambda t: torch.nn.functional.gelu(
                        t, approximate="tanh")}[name](x)
        return _Act.apply(x, opid)
    register_gpu("act_" + name)(fn)
    return fn


for _n, _i in _ACT_IDS.items():
    _make_act(_n, _i)


# ---------------------------------------------------------------------------
# fused multi-tensor optimizer hooks (used by gluon.Trainer fused path)
# ---------------------------------------------------------------------------

def build_opt_descs(ws, gs, ms, vs, w32s):
    return _C.build_opt_descs(ws, gs, ms, vs, w32s)


def multi_opt_run(blob, ntens, nchunks, mode, wbf16, lr, momentum, beta1,
                  beta2, eps, wd, rescale, clip):
    _C.multi_opt_run(blob, ntens, nchunks, mode, wbf16, lr, momentum, beta1,
                     beta2, eps, wd, rescale, clip)


def multi_sumsq(blob, ntens, nchunks, bf16in):
    return _C.multi_sumsq_run(blob, ntens, nchunks, bf16in)


# ---------------------------------------------------------------------------
# Multi-head attention (composed from the MFMA batched GEMM + masked-softmax
# kernels; reference parity: contrib/transformer.cu interleaved matmuls.
# A fully-fused flash-style kernel is the planned fast path.)
# ---------------------------------------------------------------------------

class _MaskedSoftmax(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, vlen):
        y = _C.softmax_masked_fwd(x.contiguous(), vlen, 1.0)
        ctx.save_for_backward(y)
        return y

    @staticmethod
    def backward(ctx, dy):
        (y,) = ctx.saved_tensors
        # masked cols have y == 0 -> zero grad automatically
        return _C.softmax_bwd(y, dy.contiguous(), False, 1.0), None


_causal_vlen_cache = {}


def _causal_vlen(BH, T, device):
    key = (BH, T, device)
    v = _causal_vlen_cache.get(key)
    if v is None:
        v = torch.arange(1, T + 1, dtype=torch.int32,
                         device=device).repeat(BH)
        _causal_vlen_cache[key] = v
    return v


class _MhaCore(torch.autograd.Function):
    """Zero-copy multi-head attention: heads are read from / written to the
    [B, S, H*D] activation IN PLACE by the three layout-form GEMM kernels
    (attn_scores = TN, attn_ctx = NN with tr-reads, attn_wg = WG) — no
    [B,H,S,D] permute copies anywhere in fwd or bwd, and the 1/sqrt(d)
    scale is folded into the softmax kernel."""

    @staticmethod
    def forward(ctx, q, k, v, H, causal, scale, dropout_p, training):
        q, k, v = q.contiguous(), k.contiguous(), v.contiguous()
        B, Tq, HD = q.shape
        s = _C.attn_scores(q, k, H)                  # [BH, Tq, Tk] logits
        vlen = _causal_vlen(B * H, Tq, s.device) if causal else None
        if dropout_p > 0.0 and training:
            # fused probs+dropout: one kernel yields p (softmax bwd),
            # pd (ctx/dV operand) and the mask
            p, pd, mask = _C.softmax_dropout_fwd(s, vlen, scale,
                                                 dropout_p, _next_seed())
        elif causal:
            p = _C.softmax_masked_fwd(s, vlen, scale)
            pd, mask = p, None
        else:
            p = _C.softmax_fwd(s, False, scale)
            pd, mask = p, None
        out = _C.attn_ctx(pd, v, H)                  # [B, Tq, HD]
        ctx.save_for_backward(q, k, v, p, pd, mask)
        ctx.H, ctx.scale, ctx.dropout_p = H, scale, dropout_p
        return out

    @staticmethod
    def backward(ctx, dout):
        q, k, v, p, pd, mask = ctx.saved_tensors
        H, scale, dropout_p = ctx.H, ctx.scale, ctx.dropout_p
        dout = dout.contiguous()
        dpd = _C.attn_scores(dout, v, H)             # dP (post-dropout)
        dv = _C.attn_wg(pd, dout, H)
        if mask is not None:
            # fused dropout-bwd + softmax-bwd (+ scale fold)
            ds = _C.softmax_dropout_bwd(p, dpd, mask, scale, dropout_p)
        else:
            ds = _C.softmax_bwd(p, dpd, False, scale)
        dq = _C.attn_ctx(ds, k, H)
        dk = _C.attn_wg(ds, q, H)
        return dq, dk, dv, None, None, None, None, None


class _FlashMha(torch.autograd.Function):
    """Fully-fused flash attention (flash.hip): S/P/mask never touch HBM;
    backward recomputes probabilities from the saved (m, l) stats and
    replays the dropout hash."""

    @staticmethod
    def forward(ctx, q, k, v, H, causal, scale, dropout_p, training):
        q, k, v = q.contiguous(), k.contiguous(), v.contiguous()
        dp = dropout_p if training else 0.0
        salt = _next_seed()
        out, m, l = _C.flash_fwd(q, k, v, H, scale, causal, dp, salt)
        ctx.save_for_backward(q, k, v, out, m, l)
        ctx.conf = (H, scale, causal, dp, salt)
        return out

    @staticmethod
    def backward(ctx, dout):
        q, k, v, out, m, l = ctx.saved_tensors
        H, scale, causal, dp, salt = ctx.conf
        dq, dk, dv = _C.flash_bwd(q, k, v, out, dout.contiguous(), m, l,
                                  H, scale, causal, dp, salt)
        return dq, dk, dv, None, None, None, None, None


import os as _os
_USE_FLASH = _os.environ.get("MXA_FLASH", "1") != "0"


@register_gpu("mha")
def mha(q, k, v, num_heads, causal, scale, dropout_p=0.0, training=False):
    B, Tq, HD = q.shape
    Tk = k.shape[1]
    D = HD // num_heads
    if (_USE_FLASH and D in (64, 128) and (Tq == Tk or not causal)
            and q.dtype == torch.bfloat16):
        # any sequence length (tail blocks are masked in-kernel) and
        # cross-attention Tq != Tk (decoder enc-dec attention) run fused;
        # only causal masking assumes a square score matrix
        return _FlashMha.apply(q, k, v, num_heads, causal, scale,
                               dropout_p, training)
    if (D % 64 == 0 and Tk % 64 == 0 and Tq % 8 == 0
            and q.dtype == torch.bfloat16):
        return _MhaCore.apply(q, k, v, num_heads, causal, scale,
                              dropout_p, training)
    # fallback (odd head dims / seq lens): materialized head layout
    qh = q.reshape(B, Tq, num_heads, D).transpose(1, 2) \
        .reshape(B * num_heads, Tq, D).contiguous()
    kh = k.reshape(B, Tk, num_heads, D).transpose(1, 2) \
        .reshape(B * num_heads, Tk, D).contiguous()
    vh = v.reshape(B, Tk, num_heads, D).transpose(1, 2) \
        .reshape(B * num_heads, Tk, D).contiguous()
    s = _tn_or_matmul(qh, kh) * scale            # [BH, Tq, Tk]
    if causal:
        p = _MaskedSoftmax.apply(s, _causal_vlen(B * num_heads, Tq,
                                                 s.device))
    else:
        p = _Softmax.apply(s, False)
    if dropout_p > 0.0 and training:
        p = _Dropout.apply(p, dropout_p)
    vt = vh.transpose(-1, -2).contiguous()       # [BH, D, Tk]
    out = _tn_or_matmul(p.contiguous(), vt)      # [BH, Tq, D]
    return out.reshape(B, num_heads, Tq, D).transpose(1, 2) \
        .reshape(B, Tq, HD)


# ---------------------------------------------------------------------------
# fp8 (OCP e4m3) linear path — config #5 (Transformer-big fp8).
# Forward runs the fp8 MFMA GEMM with per-tensor amax scaling; backward
# runs bf16 (standard fp8 training recipe: e4m3 fwd, higher-precision grads).
# ---------------------------------------------------------------------------

E4M3_MAX = 448.0


def _amax_scale(t):
    """Device-resident scale (no host sync — the kernels read it)."""
    amax = t.detach().abs().amax().clamp(min=1e-12).float()
    return E4M3_MAX / amax


class _Fp8Linear(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, bias):
        ctx.save_for_backward(x, w)
        ctx.has_bias = bias is not None
        # MX block-scaled fp8 (per-32-element e8m0 scales) runs on the
        # double-rate mfma_scale path; per-tensor-scaled fp8 kept as the
        # fallback for K % 128 != 0.
        if x.shape[-1] % 128 == 0:
            x8, xs = _C.mx_cast(x.contiguous())
            w8, ws = _C.mx_cast(w.contiguous())
            return _C.gemm_mx(x8, xs, w8, ws,
                              bias.float() if bias is not None else None,
                              False)
        sx = _amax_scale(x)
        sw = _amax_scale(w)
        x8 = _C.cast_to_fp8(x.contiguous(), sx)
        w8 = _C.cast_to_fp8(w.contiguous(), sw)
        return _C.gemm_tn_fp8(x8, w8,
                              bias.float() if bias is not None else None,
                              sx, sw, False)

    @staticmethod
    def backward(ctx, dy):
        x, w = ctx.saved_tensors
        dy = dy.contiguous()
        dx = dw = db = None
        M, N = dy.shape
        K = w.shape[1]
        # fp8 gradient GEMMs (MX recipe: e5m2 grads x e4m3 params/acts,
        # fp32 accumulate).  Both backward GEMMs run as plain TN MX GEMMs
        # over transpose-quantized operands:
        #   dx[M,K] = mx8(dy8 [M, N-blocks e5m2],  Wt8 [K, N-blocks e4m3])
        #   dW[N,K] = mx8(dyT8 [N, M-blocks e5m2], xT8 [K, M-blocks e4m3])
        # so no WG/NN-form MX kernels are needed.  MXA_FP8_GRADS=0 falls
        # back to the bf16 grads of round 1.
        # default OFF: measured (profiles/RESULTS.md r2) the transpose
        # casts + grad GEMMs cost more than the bf16 grads they replace
        # (198k vs 259k tok/s e2e); the shipped fp8 recipe is e4m3 fwd +
        # bf16 grads.  MXA_FP8_GRADS=1 enables the full-fp8 backward.
        use_mx = (N % 128 == 0 and M % 128 == 0 and K % 32 == 0 and
                  os.environ.get("MXA_FP8_GRADS", "0") == "1")
        if use_mx:
            if ctx.needs_input_grad[0]:
                dy8, dys = _C.mx_cast(dy, True)
                wt8, wts = _C.mx_cast_t(w, False)
                dx = _C.gemm_mx(dy8, dys, wt8, wts, None, False, 1, 0)
            if ctx.needs_input_grad[1]:
                dyt8, dyts = _C.mx_cast_t(dy, True)
                xt8, xts = _C.mx_cast_t(x, False)
                dw = _C.gemm_mx(dyt8, dyts, xt8, xts, None, False, 1, 0)
        else:
            if ctx.needs_input_grad[0]:
                if w.shape[0] % 64 == 0 and w.shape[1] % 8 == 0:
                    dx = _C.gemm_nn(dy, w)
                else:
                    dx = torch.matmul(dy, w)
            if ctx.needs_input_grad[1]:
                dw = _wg_or_matmul(dy, x, w.dtype)
        if ctx.has_bias and ctx.needs_input_grad[2]:
            db = _C.colsum(dy)
        return dx, dw, db


@register_gpu("fp8_linear")
def fp8_linear(x, w, bias):
    if x.size(-1) % 128 != 0:
        # K not tileable for the fp8 kernel: bf16 MFMA path
        return linear(x, w, bias)
    x2 = x.reshape(-1, x.shape[-1])
    out = _Fp8Linear.apply(x2.contiguous(), w,
                           bias if bias is None else bias.to(x.dtype))
    return out.reshape(x.shape[:-1] + (w.shape[0],))


# ---------------------------------------------------------------------------
# Generic elementwise engine (ew.hip) — the framework-owned long-tail op
# surface (reference mxnet_op.h Kernel<OP,gpu>::Launch).  core.py consults
# these tables for GPU bf16 tensors; everything else stays on the CPU
# oracle / torch substrate.
# ---------------------------------------------------------------------------

EW_UNARY = {
    "relu": 0, "sigmoid": 1, "tanh": 2, "exp": 3, "log": 4, "sqrt": 5,
    "rsqrt": 6, "square": 7, "abs": 8, "negative": 9, "reciprocal": 10,
    "sin": 11, "cos": 12, "erf": 13, "log1p": 14, "expm1": 15,
    "floor": 16, "ceil": 17, "round": 18, "sign": 19, "softsign": 20,
}

EW_BINARY = {"elemwise_add": 0, "broadcast_add": 0, "elemwise_sub": 1,
             "broadcast_sub": 1, "elemwise_mul": 2, "broadcast_mul": 2,
             "elemwise_div": 3, "broadcast_div": 3, "broadcast_power": 4,
             "power": 4, "broadcast_maximum": 5, "maximum": 5,
             "broadcast_minimum": 6, "minimum": 6}


class _EwUnary(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, opid):
        y = _C.ew_unary(opid, x.contiguous())
        ctx.save_for_backward(x, y)
        ctx.opid = opid
        return y

    @staticmethod
    def backward(ctx, dy):
        x, y = ctx.saved_tensors
        return _C.ew_unary_grad(ctx.opid, x.contiguous(), y,
                                dy.contiguous()), None


class _EwBinary(torch.autograd.Function):
    """Same-shape binary with analytic backwards on the same engine.
    Saves only the per-op backward operands (add/sub save NOTHING — the
    residual adds of every model go through here)."""

    @staticmethod
    def forward(ctx, a, b, opid):
        y = _C.ew_binary(opid, a.contiguous(), b.contiguous())
        ctx.opid = opid
        if opid in (0, 1):
            ctx.save_for_backward()
        elif opid == 2:
            ctx.save_for_backward(a, b)
        elif opid == 3:
            ctx.save_for_backward(b, y)
        else:
            ctx.save_for_backward(a, b, y)
        return y

    @staticmethod
    def backward(ctx, dy):
        saved = ctx.saved_tensors
        a = b = y = None
        if ctx.opid == 2:
            a, b = saved
        elif ctx.opid == 3:
            b, y = saved
        elif ctx.opid in (4, 5, 6):
            a, b, y = saved
        dy = dy.contiguous()
        op = ctx.opid
        da = db = None
        need_a, need_b = ctx.needs_input_grad[0], ctx.needs_input_grad[1]
        if op == 0:    # add
            da = dy if need_a else None
            db = dy if need_b else None
        elif op == 1:  # sub
            da = dy if need_a else None
            db = _C.ew_unary(EW_UNARY["negative"], dy) if need_b else None
        elif op == 2:  # mul
            da = _C.ew_binary(2, dy, b.contiguous()) if need_a else None
            db = _C.ew_binary(2, dy, a.contiguous()) if need_b else None
        elif op == 3:  # div
            if need_a:
                da = _C.ew_binary(3, dy, b.contiguous())
            if need_b:
                db = _C.ew_binary(
                    2, _C.ew_unary(EW_UNARY["negative"], dy),
                    _C.ew_binary(3, y, b.contiguous()))
        elif op == 4:  # pow: da = dy * b * a^(b-1); db = dy * y * ln a
            if need_a:
                da = _C.ew_binary(
                    2, dy, _C.ew_binary(
                        2, b.contiguous(),
                        _C.ew_binary(3, y, a.contiguous())))
            if need_b:
                db = _C.ew_binary(
                    2, dy, _C.ew_binary(
                        2, y, _C.ew_unary(EW_UNARY["log"],
                                          a.contiguous())))
        elif op in (5, 6):  # max/min: grad to the winning side
            amask = (a >= b) if op == 5 else (a <= b)
            if need_a:
                da = dy * amask.to(dy.dtype)
            if need_b:
                db = dy * (~amask).to(dy.dtype)
        return da, db, None


class _EwReduceLast(torch.autograd.Function):
    """sum over the last axis (fp32 accumulation), bf16 out."""

    @staticmethod
    def forward(ctx, x):
        ctx.D = x.shape[-1]
        y = _C.ew_reduce_last(0, x.contiguous())
        return y.to(x.dtype)

    @staticmethod
    def backward(ctx, dy):
        return dy.unsqueeze(-1).expand(dy.shape + (ctx.D,)).contiguous()


def ew_unary_dispatch(name, x):
    """Native launch when eligible, else None (caller falls back)."""
    opid = EW_UNARY.get(name)
    if (opid is None or not isinstance(x, torch.Tensor) or not x.is_cuda
            or x.dtype != torch.bfloat16):
        return None
    return _EwUnary.apply(x, opid)


def ew_binary_dispatch(name, a, b):
    opid = EW_BINARY.get(name)
    if (opid is None or not isinstance(a, torch.Tensor)
            or not isinstance(b, torch.Tensor) or not a.is_cuda
            or not b.is_cuda or a.dtype != torch.bfloat16
            or b.dtype != torch.bfloat16):
        return None
    if a.shape == b.shape:
        return _EwBinary.apply(a, b, opid)
    # broadcast fwd runs native when no grad is needed; the backward of a
    # broadcast needs reduction over the broadcast dims (substrate)
    if not torch.is_grad_enabled() or (not a.requires_grad
                                       and not b.requires_grad):
        if a.dim() <= 6 and b.dim() <= 6:
            return _C.ew_binary(opid, a.contiguous(), b.contiguous())
    return None


def ew_sum_last(x):
    if (not isinstance(x, torch.Tensor) or not x.is_cuda
            or x.dtype != torch.bfloat16 or x.dim() < 1
            or not x.is_contiguous()):
        return None
    return _EwReduceLast.apply(x)
