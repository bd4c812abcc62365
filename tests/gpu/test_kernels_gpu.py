"""GPU numerics tests: every HIP kernel vs a plain torch fp32 reference
(reference pattern: tests/python/gpu/test_operator_gpu.py check_consistency
— the CPU/fp32 implementation is the oracle)."""
import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from anand_mxnet_amd import _C
else:  # collection on CPU boxes; tests are skipped by conftest
    _C = None

BF = torch.bfloat16


def _dev():
    return torch.device("cuda", 0)


def _rt(shape, scale=1.0):
    return (torch.rand(shape, device=_dev(), dtype=torch.float32) * 2 - 1) \
        * scale


def assert_close(a, b, rtol=3e-2, atol=3e-2, msg=""):
    a = a.float().cpu()
    b = b.float().cpu()
    err = (a - b).abs().max().item()
    denom = b.abs().max().item() + 1e-6
    assert torch.allclose(a, b, rtol=rtol, atol=atol), \
        f"{msg}: max abs err {err:.4e} (ref max {denom:.3e})"


def test_gemm_tn_vs_fp32():
    for (M, N, K) in [(128, 128, 64), (256, 512, 192), (100, 130, 64),
                      (1024, 1000, 512)]:
        a = _rt((M, K)).to(BF).float()   # round inputs: isolates accum error
        b = _rt((N, K)).to(BF).float()
        ref = a @ b.t()
        out = _C.gemm_tn(a.to(BF).contiguous(), b.to(BF).contiguous(),
                         None, False)
        assert_close(out, ref, msg=f"gemm {M}x{N}x{K}")


def test_gemm_tn_bias_relu():
    a, b = _rt((256, 128)), _rt((64, 128))
    bias = _rt((64,))
    ref = torch.relu(a @ b.t() + bias)
    out = _C.gemm_tn(a.to(BF), b.to(BF), bias, True)
    assert_close(out, ref, msg="gemm bias+relu")


def test_gemm_tn_batched():
    a, b = _rt((4, 128, 64)), _rt((4, 96, 64))
    ref = a @ b.transpose(-1, -2)
    out = _C.gemm_tn(a.to(BF).contiguous(), b.to(BF).contiguous(), None,
                     False)
    assert_close(out, ref, msg="batched gemm")


def test_gemm_wg_vs_fp32():
    for (m, Mo, No) in [(512, 128, 128), (1000, 64, 256), (4096, 256, 104)]:
        a = _rt((m, Mo)).to(BF).float()
        b = _rt((m, No)).to(BF).float()
        ref = a.t() @ b
        out = _C.gemm_wg(a.to(BF).contiguous(), b.to(BF).contiguous())
        assert_close(out, ref, rtol=3e-2, atol=5e-2, msg=f"wg {m}x{Mo}x{No}")
        # transposed fold epilogue (the conv weight-grad permute path)
        outT = _C.gemm_wg(a.to(BF).contiguous(), b.to(BF).contiguous(),
                          True)
        assert_close(outT, ref.t(), rtol=3e-2, atol=5e-2,
                     msg=f"wgT {m}x{Mo}x{No}")


def _conv_ref(x, w, stride, pad):
    # x NHWC f32, w (K,R,S,C): torch conv in NCHW
    xc = x.permute(0, 3, 1, 2)
    wc = w.permute(0, 3, 1, 2)
    y = torch.nn.functional.conv2d(xc, wc, None, stride, pad)
    return y.permute(0, 2, 3, 1)


@pytest.mark.parametrize("cfg", [
    ((2, 16, 16, 64), 64, 3, 1, 1),     # 3x3 s1
    ((2, 16, 16, 64), 128, 3, 2, 1),    # 3x3 s2
    ((2, 14, 14, 128), 256, 1, 1, 0),   # 1x1
    ((2, 14, 14, 128), 256, 1, 2, 0),   # 1x1 s2
])
def test_conv2d_fwd(cfg):
    xs, K, R, s, p = cfg
    x = _rt(xs).to(BF).float()
    w = _rt((K, R, R, xs[3]), 0.2).to(BF).float()
    ref = _conv_ref(x, w, (s, s), (p, p))
    w_rskc = w.to(BF).permute(1, 2, 0, 3).contiguous()
    y = _C.conv2d_fwd(x.to(BF).contiguous(), w_rskc, None, [s, s], [p, p],
                      [1, 1], False)
    assert_close(y, ref, msg=f"conv fwd {cfg}")


@pytest.mark.parametrize("cfg", [
    ((2, 16, 16, 64), 64, 3, 1, 1),
    ((2, 16, 16, 64), 128, 3, 2, 1),
    ((2, 14, 14, 128), 256, 1, 2, 0),
])
def test_conv2d_dgrad_wgrad(cfg):
    xs, K, R, s, p = cfg
    x = _rt(xs).to(BF).float().requires_grad_(True)
    w = _rt((K, R, R, xs[3]), 0.2).to(BF).float().requires_grad_(True)
    y = _conv_ref(x, w, (s, s), (p, p))
    dy = _rt(tuple(y.shape)).to(BF).float()
    y.backward(dy)

    w_rsck = w.detach().to(BF).permute(1, 2, 3, 0).contiguous()
    dx = _C.conv2d_dgrad(dy.to(BF).contiguous(), w_rsck, list(xs), [s, s],
                         [p, p], [1, 1])
    assert_close(dx, x.grad, msg=f"conv dgrad {cfg}")

    dwf = _C.conv2d_wgrad(x.detach().to(BF).contiguous(),
                          dy.to(BF).contiguous(), [R, R], [s, s], [p, p],
                          [1, 1])  # (K,R,S,C) bf16 (fold epilogue)
    assert_close(dwf, w.grad, rtol=3e-2, atol=5e-2, msg=f"conv wgrad {cfg}")


def test_bn_train_fwd_bwd():
    N, H, W, C = 4, 8, 8, 64
    x = _rt((N, H, W, C))
    gamma = torch.rand(C, device=_dev()) + 0.5
    beta = _rt((C,))
    rmean = torch.zeros(C, device=_dev())
    rvar = torch.ones(C, device=_dev())
    xt = x.clone().requires_grad_(True)
    ref = torch.nn.functional.batch_norm(
        xt.permute(0, 3, 1, 2), None, None, gamma, beta, True, 0.1,
        1e-5).permute(0, 2, 3, 1)
    y, sm, sr = _C.bn_fwd_train(x.to(BF).contiguous(), gamma, beta, rmean,
                                rvar, 1e-5, 0.9, False)
    assert_close(y, ref, msg="bn fwd")
    # bwd
    dy = _rt((N, H, W, C))
    ref.backward(dy)
    dx, dg, db = _C.bn_bwd(x.to(BF).contiguous(), dy.to(BF).contiguous(),
                           gamma, beta, sm, sr, False)
    assert_close(dx, xt.grad, rtol=5e-2, atol=5e-2, msg="bn dx")
    # running stats: mxnet momentum 0.9 == torch momentum 0.1
    ref_mean = 0.1 * x.reshape(-1, C).mean(0)
    assert_close(rmean, ref_mean, rtol=2e-2, atol=2e-3, msg="bn rmean")


def test_bn_relu_fused():
    N, H, W, C = 2, 4, 4, 64
    x = _rt((N, H, W, C))
    gamma = torch.ones(C, device=_dev())
    beta = torch.zeros(C, device=_dev())
    rmean = torch.zeros(C, device=_dev())
    rvar = torch.ones(C, device=_dev())
    y, sm, sr = _C.bn_fwd_train(x.to(BF).contiguous(), gamma, beta, rmean,
                                rvar, 1e-5, 0.9, True)
    assert (y.float() >= 0).all()


def test_maxpool():
    x = _rt((2, 16, 16, 64)).to(BF).float()  # same quantization as kernel
    xt = x.clone().requires_grad_(True)
    ref = torch.nn.functional.max_pool2d(xt.permute(0, 3, 1, 2), 3, 2,
                                         1).permute(0, 2, 3, 1)
    y, arg = _C.maxpool(x.to(BF).contiguous(), [3, 3], [2, 2], [1, 1])
    assert_close(y, ref, msg="maxpool fwd")
    dy = _rt(tuple(ref.shape))
    ref.backward(dy)
    dx = _C.maxpool_bwd(dy.to(BF).contiguous(), arg, list(x.shape), [3, 3],
                        [2, 2], [1, 1])
    assert_close(dx, xt.grad, rtol=5e-2, atol=5e-2, msg="maxpool bwd")


def test_global_avgpool():
    x = _rt((3, 7, 7, 128))
    ref = x.mean(dim=(1, 2), keepdim=True)
    y = _C.global_avgpool(x.to(BF).contiguous())
    assert_close(y, ref, msg="gap fwd")
    dy = _rt((3, 1, 1, 128))
    dx = _C.global_avgpool_bwd(dy.to(BF).contiguous(), list(x.shape))
    ref_dx = (dy / 49.0).expand(3, 7, 7, 128)
    assert_close(dx, ref_dx, msg="gap bwd")


def test_softmax_fwd_bwd():
    x = _rt((64, 128), 4.0)
    xt = x.clone().requires_grad_(True)
    ref = torch.softmax(xt, dim=-1)
    y = _C.softmax_fwd(x.to(BF).contiguous(), False)
    assert_close(y, ref, rtol=2e-2, atol=2e-3, msg="softmax fwd")
    dy = _rt((64, 128))
    ref.backward(dy)
    dx = _C.softmax_bwd(y, dy.to(BF).contiguous(), False)
    assert_close(dx, xt.grad, rtol=5e-2, atol=1e-2, msg="softmax bwd")


def test_log_softmax():
    x = _rt((32, 1000), 3.0)
    # pad to %8 is satisfied (1000%8==0)
    ref = torch.log_softmax(x, dim=-1)
    y = _C.softmax_fwd(x.to(BF).contiguous(), True)
    assert_close(y, ref, rtol=2e-2, atol=3e-2, msg="log_softmax")


def test_layernorm_fwd_bwd():
    x = _rt((128, 768), 2.0)
    gamma = torch.rand(768, device=_dev()) + 0.5
    beta = _rt((768,))
    xt = x.clone().requires_grad_(True)
    gt = gamma.clone().requires_grad_(True)
    bt = beta.clone().requires_grad_(True)
    ref = torch.nn.functional.layer_norm(xt, (768,), gt, bt, 1e-5)
    y, mean, rstd = _C.layernorm_fwd(x.to(BF).contiguous(), gamma, beta,
                                     1e-5)
    assert_close(y, ref, msg="ln fwd")
    dy = _rt((128, 768))
    ref.backward(dy)
    dx, dg, db = _C.layernorm_bwd(x.to(BF).contiguous(),
                                  dy.to(BF).contiguous(), gamma, mean, rstd)
    assert_close(dx, xt.grad, rtol=5e-2, atol=5e-2, msg="ln dx")
    assert_close(dg, gt.grad, rtol=5e-2, atol=5e-1, msg="ln dgamma")
    assert_close(db, bt.grad, rtol=5e-2, atol=5e-1, msg="ln dbeta")


def test_dropout():
    x = torch.ones((1024, 64), device=_dev(), dtype=BF)
    y, mask = _C.dropout_fwd(x, 0.5, 1234)
    keep = mask.float().mean().item()
    assert 0.4 < keep < 0.6
    yv = y.float()
    assert ((yv == 0) | ((yv - 2.0).abs() < 1e-2)).all()
    dy = torch.ones_like(x)
    dx = _C.dropout_bwd(dy, mask, 0.5)
    assert_close(dx, y, msg="dropout bwd == fwd pattern for ones")


def test_embedding():
    W = _rt((1000, 64))
    idx = torch.randint(0, 1000, (4, 37), device=_dev())
    ref = torch.nn.functional.embedding(idx, W)
    out = _C.embedding_fwd(idx, W.to(BF).contiguous())
    assert_close(out, ref, msg="embedding fwd")
    dy = _rt((4, 37, 64))
    dW = _C.embedding_bwd(idx, dy.to(BF).contiguous(), 1000)
    ref_dW = torch.zeros_like(W)
    ref_dW.index_add_(0, idx.flatten(), dy.reshape(-1, 64))
    assert_close(dW, ref_dW, rtol=3e-2, atol=3e-2, msg="embedding bwd")


def test_activations():
    x = _rt((512, 64), 2.0)
    for name, opid, fn in [
        ("relu", 0, torch.relu), ("sigmoid", 1, torch.sigmoid),
        ("tanh", 2, torch.tanh),
        ("gelu", 3, torch.nn.functional.gelu),
        ("silu", 4, torch.nn.functional.silu),
    ]:
        xt = x.clone().requires_grad_(True)
        ref = fn(xt)
        y = _C.act_fwd(opid, x.to(BF).contiguous())
        assert_close(y, ref, msg=f"act {name} fwd")
        dy = _rt((512, 64))
        ref.backward(dy)
        dx = _C.act_bwd(opid, x.to(BF).contiguous(), dy.to(BF).contiguous())
        assert_close(dx, xt.grad, rtol=5e-2, atol=2e-2,
                     msg=f"act {name} bwd")


def test_multi_opt_sgd_mom():
    torch.manual_seed(0)
    shapes = [(64, 64), (1000,), (3, 3, 64, 64)]
    ws32 = [torch.randn(s, device=_dev()) for s in shapes]
    gs32 = [torch.randn(s, device=_dev()) for s in shapes]
    ws = [w.to(BF) for w in ws32]
    gs = [g.to(BF) for g in gs32]
    moms = [torch.zeros(s, device=_dev()) for s in shapes]
    w32s = [w.float().clone() for w in ws]
    blob, meta = _C.build_opt_descs(ws, gs, moms, [], w32s)
    ntens, nchunks = int(meta[0]), int(meta[1])
    lr, mom, wd = 0.1, 0.9, 1e-4
    _C.multi_opt_run(blob, ntens, nchunks, 1, True, lr, mom, 0.9, 0.999,
                     1e-8, wd, 1.0, -1.0)
    # reference recompute from scratch
    for i, s in enumerate(shapes):
        w0 = ws32[i].to(BF).float()
        g = gs[i].float()
        m = -lr * (g + wd * w0)
        ref = w0 + m
        assert_close(w32s[i], ref, rtol=1e-4, atol=1e-4,
                     msg=f"mp_sgd tensor {i}")
        assert_close(ws[i].float(), ref, rtol=1e-2, atol=1e-2)


def test_multi_sumsq():
    ws = [torch.randn((100,), device=_dev()).to(BF),
          torch.randn((70000,), device=_dev()).to(BF)]
    blob, meta = _C.build_opt_descs(ws, ws, [], [], [])
    out = _C.multi_sumsq_run(blob, int(meta[0]), int(meta[1]), True)
    for i, w in enumerate(ws):
        ref = w.float().pow(2).sum()
        assert_close(out[i], ref, rtol=1e-2, atol=1e-2, msg=f"sumsq {i}")


def test_gemm_fp8():
    M, N, K = 256, 192, 256
    a = _rt((M, K))
    b = _rt((N, K), 0.3)
    E4 = 448.0
    sa = (E4 / a.abs().max()).float()
    sb = (E4 / b.abs().max()).float()
    a8 = _C.cast_to_fp8(a.to(BF).contiguous(), sa)
    b8 = _C.cast_to_fp8(b.to(BF).contiguous(), sb)
    out = _C.gemm_tn_fp8(a8, b8, None, sa, sb, False)
    ref = a @ b.t()
    # fp8 e4m3 has ~2 mantissa bits: loose elementwise, tight correlation
    import numpy as np
    o, r = out.float().cpu().numpy(), ref.cpu().numpy()
    corr = np.corrcoef(o.ravel(), r.ravel())[0, 1]
    assert corr > 0.995, corr
    rel = np.abs(o - r).max() / (np.abs(r).max() + 1e-6)
    assert rel < 0.15, rel


def test_cast_fp8_roundtrip():
    x = torch.linspace(-400, 400, 1024, device=_dev()).to(BF)
    y8 = _C.cast_to_fp8(x.contiguous(), torch.ones((), device=_dev()))
    ref = x.float().to(torch.float8_e4m3fn).view(torch.uint8)
    assert (y8 == ref).float().mean().item() > 0.99


def test_gemm_nn_vs_fp32():
    for (M, N, K) in [(128, 128, 64), (256, 512, 192), (100, 136, 64),
                      (512, 1000, 256)]:
        if N % 8:
            continue
        a = _rt((M, K)).to(BF).float()
        b = _rt((K, N)).to(BF).float()
        ref = a @ b
        out = _C.gemm_nn(a.to(BF).contiguous(), b.to(BF).contiguous())
        assert_close(out, ref, msg=f"gemm_nn {M}x{N}x{K}")


def test_colsum():
    for (m, N) in [(1024, 128), (8192, 768), (777, 1000)]:
        x = _rt((m, N)).to(BF)
        ref = x.float().sum(0)
        out = _C.colsum(x.contiguous())
        assert_close(out, ref, rtol=1e-3, atol=1e-2, msg=f"colsum {m}x{N}")


def _heads(x, H):
    B, T, HD = x.shape
    return x.reshape(B, T, H, HD // H).transpose(1, 2)  # [B,H,T,D]


def test_attn_scores_ctx_wg():
    B, H, Tq, Tk, D = 3, 4, 128, 192, 64
    q = _rt((B, Tq, H * D)).to(BF)
    k = _rt((B, Tk, H * D)).to(BF)
    v = _rt((B, Tk, H * D)).to(BF)
    s = _C.attn_scores(q.contiguous(), k.contiguous(), H)
    ref_s = (_heads(q, H).float() @ _heads(k, H).float().transpose(-1, -2)) \
        .reshape(B * H, Tq, Tk)
    assert_close(s, ref_s, msg="attn_scores")

    p = torch.nn.functional.softmax(ref_s.to(BF).float(), dim=-1).to(BF)
    o = _C.attn_ctx(p.contiguous(), v.contiguous(), H)
    ref_o = (p.float().reshape(B, H, Tq, Tk) @ _heads(v, H).float()) \
        .transpose(1, 2).reshape(B, Tq, H * D)
    assert_close(o, ref_o, msg="attn_ctx")

    dk = _C.attn_wg(p.contiguous(), q.contiguous(), H)
    ref_dk = (p.float().reshape(B, H, Tq, Tk).transpose(-1, -2)
              @ _heads(q, H).float()).transpose(1, 2).reshape(B, Tk, H * D)
    assert_close(dk, ref_dk, msg="attn_wg")


def test_mha_fwd_bwd_vs_torch():
    from anand_mxnet_amd.ops.hip import mha
    B, H, T, D = 2, 4, 128, 64
    scale = 1.0 / (D ** 0.5)
    for causal in (False, True):
        q0 = _rt((B, T, H * D)).to(BF)
        k0 = _rt((B, T, H * D)).to(BF)
        v0 = _rt((B, T, H * D)).to(BF)
        q1 = q0.clone().requires_grad_(True)
        k1 = k0.clone().requires_grad_(True)
        v1 = v0.clone().requires_grad_(True)
        out = mha(q1, k1, v1, H, causal, scale, 0.0, True)
        g = _rt((B, T, H * D)).to(BF)
        out.backward(g)

        q2 = q0.clone().float().requires_grad_(True)
        k2 = k0.clone().float().requires_grad_(True)
        v2 = v0.clone().float().requires_grad_(True)
        qh, kh, vh = _heads(q2, H), _heads(k2, H), _heads(v2, H)
        sc = qh @ kh.transpose(-1, -2) * scale
        if causal:
            mask = torch.triu(torch.ones(T, T, device=_dev(), dtype=torch.bool),
                              diagonal=1)
            sc = sc.masked_fill(mask, float("-inf"))
        pr = torch.nn.functional.softmax(sc, dim=-1)
        ref = (pr @ vh).transpose(1, 2).reshape(B, T, H * D)
        ref.backward(g.float())

        assert_close(out, ref, rtol=5e-2, atol=5e-2, msg=f"mha out c={causal}")
        for name, got, want in (("dq", q1.grad, q2.grad),
                                ("dk", k1.grad, k2.grad),
                                ("dv", v1.grad, v2.grad)):
            cos = torch.nn.functional.cosine_similarity(
                got.float().flatten(), want.float().flatten(), dim=0).item()
            assert cos > 0.9995, (name, causal, cos)
            assert_close(got, want, rtol=6e-2, atol=6e-2,
                         msg=f"mha {name} c={causal}")


def test_im2col_pad_stem():
    N, H, W, C, R, S = 2, 20, 20, 3, 7, 7
    sh = sw = 2
    ph = pw = 3
    x = _rt((N, H, W, C)).to(BF)
    kdim = R * S * C
    kpadded = ((kdim + 63) // 64) * 64
    out = _C.im2col_pad(x.contiguous(), [R, S], [sh, sw], [ph, pw], [1, 1],
                        kpadded)
    # torch reference via unfold on NCHW
    xn = x.float().permute(0, 3, 1, 2)
    cols = torch.nn.functional.unfold(xn, (R, S), padding=(ph, pw),
                                      stride=(sh, sw))  # [N, C*R*S, P*Q]
    P = (H + 2 * ph - R) // sh + 1
    # unfold k order is (c, r, s); ours is (r, s, c)
    ref = cols.reshape(N, C, R * S, -1).permute(0, 3, 2, 1) \
        .reshape(-1, R * S * C)
    assert_close(out[:, :kdim], ref, msg="im2col_pad")
    assert out[:, kdim:].abs().max().item() == 0.0


def test_bn_add_relu_fused():
    from anand_mxnet_amd.ops.hip import batchnorm_nhwc
    N, H, W, C = 4, 8, 8, 64
    x0 = _rt((N, H, W, C)).to(BF)
    r0 = _rt((N, H, W, C)).to(BF)
    gamma = torch.rand(C, device=_dev()) + 0.5
    beta = torch.rand(C, device=_dev()) - 0.5
    rm = torch.zeros(C, device=_dev())
    rv = torch.ones(C, device=_dev())

    x1 = x0.clone().requires_grad_(True)
    r1 = r0.clone().requires_grad_(True)
    y = batchnorm_nhwc(x1, gamma, beta, rm.clone(), rv.clone(), 1e-5, 0.9,
                       True, False, r1)
    g = _rt((N, H, W, C)).to(BF)
    y.backward(g)

    x2 = x0.clone().float().requires_grad_(True)
    r2 = r0.clone().float().requires_grad_(True)
    xn = x2.permute(0, 3, 1, 2)
    yn = torch.nn.functional.batch_norm(xn, rm.clone(), rv.clone(), gamma,
                                        beta, True, 0.1, 1e-5)
    ref = torch.relu(yn.permute(0, 2, 3, 1) + r2)
    ref.backward(g.float())

    assert_close(y, ref, msg="bn_add_relu fwd")
    assert_close(x1.grad, x2.grad, rtol=5e-2, atol=5e-2, msg="bn_add dx")
    assert_close(r1.grad, r2.grad, rtol=5e-2, atol=5e-2, msg="bn_add dr")


@pytest.mark.parametrize("cfg", [
    ((2, 16, 16, 64), 128, 3, 2, 1),    # 3x3 s2: class-decomposed dgrad
    ((2, 14, 14, 128), 256, 1, 1, 0),   # 1x1 s1: plain-GEMM routing
    ((2, 14, 14, 128), 256, 1, 2, 0),   # 1x1 s2: dense-subgrid dgrad
    ((2, 16, 16, 64), 64, 3, 1, 1),     # 3x3 s1: implicit-GEMM path
])
def test_conv_op_autograd_paths(cfg):
    """The Python conv op's specialized dgrad/wgrad routings vs torch."""
    from anand_mxnet_amd.ops.hip import conv2d_nhwc
    xs, K, R, s, p = cfg
    x0 = _rt(xs).to(BF)
    w0 = _rt((K, R, R, xs[3]), 0.2).to(BF)

    x1 = x0.clone().requires_grad_(True)
    w1 = w0.clone().requires_grad_(True)
    y1 = conv2d_nhwc(x1, w1, None, (s, s), (p, p), (1, 1), 1)
    g = _rt(tuple(y1.shape)).to(BF)
    y1.backward(g)

    x2 = x0.clone().float().requires_grad_(True)
    w2 = w0.clone().float().requires_grad_(True)
    y2 = _conv_ref(x2, w2, (s, s), (p, p))
    y2.backward(g.float())

    assert_close(y1, y2, msg=f"conv op fwd {cfg}")
    assert_close(x1.grad, x2.grad, rtol=4e-2, atol=5e-2,
                 msg=f"conv op dx {cfg}")
    assert_close(w1.grad, w2.grad, rtol=4e-2, atol=6e-2,
                 msg=f"conv op dw {cfg}")


def test_softmax_dropout_fused():
    rows, D = 512, 512
    x = _rt((rows, D)).to(BF)
    p, pd, mask = _C.softmax_dropout_fwd(x.contiguous(), None, 0.5, 0.25,
                                         1234)
    ref_p = torch.nn.functional.softmax(x.float() * 0.5, dim=-1)
    assert_close(p, ref_p, msg="fused p")
    m = mask.reshape(rows, D).bool()
    keep = m.float().mean().item()
    assert abs(keep - 0.75) < 0.02, keep
    ref_pd = torch.where(m, p.float() / 0.75, torch.zeros(()).cuda())
    assert_close(pd, ref_pd, rtol=4e-2, atol=1e-3, msg="fused pd")
    # backward vs composed reference
    dpd = _rt((rows, D)).to(BF)
    ds = _C.softmax_dropout_bwd(p, dpd.contiguous(), mask, 0.5, 0.25)
    dp = torch.where(m, dpd.float() / 0.75, torch.zeros(()).cuda())
    dot = (dp * p.float()).sum(-1, keepdim=True)
    ref_ds = (dp - dot) * p.float() * 0.5
    assert_close(ds, ref_ds, rtol=4e-2, atol=2e-3, msg="fused ds")


def test_mha_dropout_path_trains():
    from anand_mxnet_amd.ops.hip import mha
    B, H, T, D = 2, 4, 128, 64
    q = _rt((B, T, H * D)).to(BF).requires_grad_(True)
    k = _rt((B, T, H * D)).to(BF).requires_grad_(True)
    v = _rt((B, T, H * D)).to(BF).requires_grad_(True)
    out = mha(q, k, v, H, False, 0.125, 0.2, True)
    out.sum().backward()
    for t in (q, k, v):
        assert t.grad is not None and torch.isfinite(t.grad.float()).all()


def test_mx_fp8_gemm():
    M, N, K = 256, 512, 768
    a = _rt((M, K)).to(BF)
    b = _rt((N, K)).to(BF)
    a8, asc = _C.mx_cast(a.contiguous())
    b8, bsc = _C.mx_cast(b.contiguous())
    out = _C.gemm_mx(a8, asc, b8, bsc)
    ref = a.float() @ b.float().t()
    # fp8 e4m3 block quantization: per-product rms err ~0.7%, dot error
    # std ~ sqrt(K)*0.007 ~ 0.19 here; bound mean/max accordingly and
    # check correlation (the decisive signal for a layout bug)
    err = (out.float() - ref).abs()
    assert err.mean().item() < 0.5, err.mean().item()
    assert err.max().item() < 2.5, err.max().item()
    cos = torch.nn.functional.cosine_similarity(
        out.float().flatten(), ref.flatten(), dim=0).item()
    assert cos > 0.999, cos


def test_mx_cast_roundtrip():
    x = _rt((64, 128)).to(BF)
    q, sc = _C.mx_cast(x.contiguous())
    assert q.shape == (64, 128) and sc.shape == (64, 4)
    # reconstruct and compare
    e = sc.int().float() - 127
    scale = torch.pow(2.0, e).repeat_interleave(32, dim=1)
    xf8 = q.view(torch.float8_e4m3fn).float()
    rec = xf8 * scale
    err = (rec - x.float()).abs().max().item()
    amax = x.float().abs().max().item()
    assert err < 0.08 * amax, (err, amax)


def test_gemm_shape_fuzz():
    """Randomized shape sweep over the three GEMM forms (dispatch edges:
    tile tails, tn8 threshold, split-K, 64-wide NN)."""
    import random
    rng = random.Random(11)
    for _ in range(12):
        M = rng.choice([8, 24, 100, 128, 129, 256, 300, 512])
        N = rng.choice([8, 16, 64, 72, 128, 200, 256, 512])
        K = rng.choice([64, 128, 192, 512])
        a = _rt((M, K)).to(BF).float()
        b = _rt((N, K)).to(BF).float()
        out = _C.gemm_tn(a.to(BF).contiguous(), b.to(BF).contiguous(),
                         None, False)
        assert_close(out, a @ b.t(), msg=f"tn {M}x{N}x{K}")
        if N % 8 == 0:
            bn = _rt((K, N)).to(BF).float()
            out2 = _C.gemm_nn(a.to(BF).contiguous(),
                              bn.to(BF).contiguous())
            assert_close(out2, a @ bn, msg=f"nn {M}x{N}x{K}")
        if M % 8 == 0 and N % 8 == 0:
            aw = _rt((K, M)).to(BF).float()
            bw = _rt((K, N)).to(BF).float()
            outw = _C.gemm_wg(aw.to(BF).contiguous(),
                              bw.to(BF).contiguous())
            assert_close(outw, aw.t() @ bw, rtol=3e-2, atol=5e-2,
                         msg=f"wg {M}x{N}x{K}")


def test_conv_shape_fuzz():
    import random
    rng = random.Random(5)
    for _ in range(6):
        C = rng.choice([64, 128])
        K = rng.choice([64, 128, 192])
        H = rng.choice([7, 9, 14, 17])
        R = rng.choice([1, 3])
        s_ = rng.choice([1, 2])
        p_ = 1 if R == 3 else 0
        N = rng.choice([1, 2, 3])
        x = _rt((N, H, H, C)).to(BF).float()
        w = _rt((K, R, R, C), 0.2).to(BF).float()
        ref = _conv_ref(x, w, (s_, s_), (p_, p_))
        w_rskc = w.to(BF).permute(1, 2, 0, 3).contiguous()
        y = _C.conv2d_fwd(x.to(BF).contiguous(), w_rskc, None,
                          [s_, s_], [p_, p_], [1, 1], False)
        assert_close(y, ref, msg=f"conv fuzz {N}x{H}x{C}->K{K} r{R}s{s_}")


def test_depthwise_conv_fwd_bwd():
    N, H, W, C, R = 2, 14, 14, 32, 3
    for s_ in (1, 2):
        x0 = _rt((N, H, W, C)).to(BF)
        w0 = _rt((C, R, R, 1), 0.3).to(BF)
        x1 = x0.clone().requires_grad_(True)
        w1 = w0.clone().requires_grad_(True)
        from anand_mxnet_amd.ops.hip import conv2d_nhwc
        y1 = conv2d_nhwc(x1, w1, None, (s_, s_), (1, 1), (1, 1), C)
        g = _rt(tuple(y1.shape)).to(BF)
        y1.backward(g)

        x2 = x0.clone().float().requires_grad_(True)
        w2 = w0.clone().float().requires_grad_(True)
        xn = x2.permute(0, 3, 1, 2)
        wn = w2.permute(0, 3, 1, 2)  # (C,1,R,R)
        y2 = torch.nn.functional.conv2d(xn, wn, None, (s_, s_), (1, 1),
                                        groups=C).permute(0, 2, 3, 1)
        y2.backward(g.float())
        assert_close(y1, y2, msg=f"dw fwd s{s_}")
        assert_close(x1.grad, x2.grad, rtol=4e-2, atol=5e-2,
                     msg=f"dw dx s{s_}")
        assert_close(w1.grad, w2.grad, rtol=4e-2, atol=6e-2,
                     msg=f"dw dw s{s_}")


def test_flash_fwd_vs_torch():
    B, H, T, D = 2, 3, 256, 64
    scale = 1.0 / (D ** 0.5)
    for causal in (False, True):
        q = _rt((B, T, H * D)).to(BF)
        k = _rt((B, T, H * D)).to(BF)
        v = _rt((B, T, H * D)).to(BF)
        out, m, l = _C.flash_fwd(q.contiguous(), k.contiguous(),
                                 v.contiguous(), H, scale, causal, 0.0, 0)
        qh = _heads(q, H).float()
        kh = _heads(k, H).float()
        vh = _heads(v, H).float()
        s = qh @ kh.transpose(-1, -2) * scale
        if causal:
            mask = torch.triu(torch.ones(T, T, device=_dev(),
                                         dtype=torch.bool), 1)
            s = s.masked_fill(mask, float("-inf"))
        p = torch.nn.functional.softmax(s, dim=-1)
        ref = (p @ vh).transpose(1, 2).reshape(B, T, H * D)
        assert_close(out, ref, rtol=5e-2, atol=5e-2,
                     msg=f"flash out c={causal}")
        # stats: l = sum exp(s - m) rowwise
        ref_m = s.max(-1).values.reshape(B * H, T)
        ref_l = torch.exp(s - ref_m.reshape(B, H, T, 1)).sum(-1) \
            .reshape(B * H, T)
        assert_close(m, ref_m, rtol=1e-2, atol=1e-2, msg="flash m")
        assert_close(l, ref_l, rtol=2e-2, atol=1e-2, msg="flash l")


def test_flash_fwd_dropout_stats():
    B, H, T, D = 1, 2, 128, 64
    q = _rt((B, T, H * D)).to(BF)
    k = _rt((B, T, H * D)).to(BF)
    v = _rt((B, T, H * D)).to(BF)
    o0, m0, l0 = _C.flash_fwd(q.contiguous(), k.contiguous(),
                              v.contiguous(), H, 0.125, False, 0.0, 7)
    o1, m1, l1 = _C.flash_fwd(q.contiguous(), k.contiguous(),
                              v.contiguous(), H, 0.125, False, 0.3, 7)
    # stats are dropout-independent; outputs differ
    assert_close(m0, m1, rtol=1e-5, atol=1e-5, msg="m stable")
    assert_close(l0, l1, rtol=1e-5, atol=1e-5, msg="l stable")
    d = (o0.float() - o1.float()).abs().mean().item()
    assert d > 1e-3, d
    # E[dropout(p)] = p: magnitudes stay comparable (signed means of
    # random attention outputs are ~0 and unusable as a ratio)
    r = (o1.float().abs().mean() / o0.float().abs().mean()).item()
    assert 0.6 < r < 1.8, r


def test_flash_bwd_vs_torch():
    B, H, T, D = 2, 3, 256, 64
    scale = 1.0 / (D ** 0.5)
    for causal in (False, True):
        q0 = _rt((B, T, H * D)).to(BF)
        k0 = _rt((B, T, H * D)).to(BF)
        v0 = _rt((B, T, H * D)).to(BF)
        g = _rt((B, T, H * D)).to(BF)

        out, m, l = _C.flash_fwd(q0.contiguous(), k0.contiguous(),
                                 v0.contiguous(), H, scale, causal, 0.0, 0)
        dq, dk, dv = _C.flash_bwd(q0.contiguous(), k0.contiguous(),
                                  v0.contiguous(), out, g.contiguous(),
                                  m, l, H, scale, causal, 0.0, 0)

        q2 = q0.clone().float().requires_grad_(True)
        k2 = k0.clone().float().requires_grad_(True)
        v2 = v0.clone().float().requires_grad_(True)
        qh, kh, vh = _heads(q2, H), _heads(k2, H), _heads(v2, H)
        s = qh @ kh.transpose(-1, -2) * scale
        if causal:
            mask = torch.triu(torch.ones(T, T, device=_dev(),
                                         dtype=torch.bool), 1)
            s = s.masked_fill(mask, float("-inf"))
        p = torch.nn.functional.softmax(s, dim=-1)
        ref = (p @ vh).transpose(1, 2).reshape(B, T, H * D)
        ref.backward(g.float())

        for name, got, want in (("dq", dq, q2.grad), ("dk", dk, k2.grad),
                                ("dv", dv, v2.grad)):
            cos = torch.nn.functional.cosine_similarity(
                got.float().flatten(), want.flatten(), dim=0).item()
            assert cos > 0.9995, (name, causal, cos)
            assert_close(got, want, rtol=6e-2, atol=6e-2,
                         msg=f"flash {name} c={causal}")


def test_flash_mha_autograd_dropout_trains():
    from anand_mxnet_amd.ops.hip import mha
    B, H, T, D = 2, 4, 128, 64
    q = _rt((B, T, H * D)).to(BF).requires_grad_(True)
    k = _rt((B, T, H * D)).to(BF).requires_grad_(True)
    v = _rt((B, T, H * D)).to(BF).requires_grad_(True)
    out = mha(q, k, v, H, True, 0.125, 0.1, True)
    out.sum().backward()
    for t in (q, k, v):
        assert t.grad is not None and torch.isfinite(t.grad.float()).all()


@pytest.mark.gpu
def test_conv_bn_fused_stats():
    """conv2d_fwd_stats / gemm_tn_stats sums must equal the direct
    per-channel sums of the conv output, and bn_fwd_train(psums=...)
    must match plain bn_fwd_train."""
    from anand_mxnet_amd import _C
    torch.manual_seed(5)
    dev = "cuda:0"
    # implicit-GEMM 3x3 route
    x = (torch.rand(32, 28, 28, 128, device=dev) * 2 - 1).to(torch.bfloat16)
    w = (torch.rand(3, 3, 128, 128, device=dev) * 0.1).to(torch.bfloat16)
    res = _C.conv2d_fwd_stats(x, w, None, [1, 1], [1, 1], [1, 1], False)
    y = res[0]
    assert len(res) == 2, "unsplit conv route should fill stats"
    stats = res[1].reshape(-1, 2, 128).sum(0)  # fold atomic slots
    y2 = y.reshape(-1, 128).float()
    ref_s = y2.sum(0)
    ref_q = (y2 * y2).sum(0)
    assert torch.allclose(stats[0], ref_s, rtol=1e-3, atol=4.0)
    assert torch.allclose(stats[1], ref_q, rtol=1e-3, atol=8.0)
    # pointwise route via gemm_tn_stats
    a = (torch.rand(4096, 256, device=dev) * 2 - 1).to(torch.bfloat16)
    b = (torch.rand(128, 256, device=dev) * 2 - 1).to(torch.bfloat16)
    res = _C.gemm_tn_stats(a, b, None, False)
    assert len(res) == 2
    o2 = res[0].float()
    st = res[1].reshape(-1, 2, 128).sum(0)
    assert torch.allclose(st[0], o2.sum(0), rtol=1e-3, atol=4.0)
    assert torch.allclose(st[1], (o2 * o2).sum(0), rtol=1e-3, atol=8.0)
    # bn consuming psums == bn computing its own stats
    g = torch.rand(128, device=dev) + 0.5
    bt = torch.rand(128, device=dev) - 0.5
    rm1, rv1 = torch.zeros(128, device=dev), torch.ones(128, device=dev)
    rm2, rv2 = torch.zeros(128, device=dev), torch.ones(128, device=dev)
    stats = _C.conv2d_fwd_stats(x, w, None, [1, 1], [1, 1], [1, 1],
                                False)[1]
    out_a = _C.bn_fwd_train(y, g, bt, rm1, rv1, 1e-5, 0.9, True, None, None)
    out_b = _C.bn_fwd_train(y, g, bt, rm2, rv2, 1e-5, 0.9, True, None,
                            stats)
    assert torch.allclose(out_a[0].float(), out_b[0].float(),
                          rtol=1e-2, atol=1e-2)
    assert torch.allclose(rm1, rm2, rtol=1e-4, atol=1e-4)
    assert torch.allclose(rv1, rv2, rtol=1e-4, atol=1e-4)


@pytest.mark.gpu
def test_conv_bn_stats_plumbing():
    """The gluon Conv2D -> BatchNorm stats handoff: a marked conv leaves
    its fused per-channel sums on the output tensor, the BN consumes
    them, and the result matches the unfused path on a single layer
    (deterministic — no deep bf16 chaos)."""
    import numpy as np
    import anand_mxnet_amd as mx
    import anand_mxnet_amd.ops.hip as _hip
    from anand_mxnet_amd import autograd, nd

    def run(flag):
        old = _hip._CONV_BN_STATS
        _hip._CONV_BN_STATS = flag
        try:
            torch.manual_seed(3)
            np.random.seed(3)
            net = mx.gluon.nn.HybridSequential()
            conv = mx.gluon.nn.Conv2D(64, 3, padding=1, use_bias=False,
                                      layout="NHWC")
            conv._collect_bn_stats = True
            net.add(conv)
            net.add(mx.gluon.nn.BatchNorm(axis=3))
            net.initialize(ctx=mx.gpu(0))
            net.cast("bfloat16")
            x = nd.array(np.random.uniform(-1, 1, (32, 28, 28, 64)),
                         ctx=mx.gpu(0), dtype="bfloat16")
            with autograd.record():
                y = net(x)
                loss = (y.astype("float32") ** 2).mean()
            loss.backward()
            g = conv.weight.grad()
            rv = net._children["1"].running_var.data()
            return (y.asnumpy().astype(np.float32),
                    g.asnumpy().astype(np.float32),
                    rv.asnumpy().astype(np.float32))
        finally:
            _hip._CONV_BN_STATS = old

    y1, g1, rv1 = run(True)
    y2, g2, rv2 = run(False)
    # running_var updated from the fused sums must match the stats-pass
    # value tightly (both f32 reductions of the same conv output)
    assert np.allclose(rv1, rv2, rtol=1e-3, atol=1e-4)
    assert np.allclose(y1, y2, rtol=5e-2, atol=5e-2)
    assert np.allclose(g1, g2, rtol=5e-2, atol=1e-3)


@pytest.mark.gpu
def test_grouped_conv_nhwc():
    """General grouped conv (ResNeXt-style) via per-group slices of the
    dense implicit-GEMM kernels, vs the torch fp32 oracle."""
    from anand_mxnet_amd.ops.registry import gpu_impl
    torch.manual_seed(0)
    dev = "cuda:0"
    N, H, W, C, K, G = 4, 14, 14, 128, 128, 2
    x = (torch.rand(N, H, W, C, device=dev) * 2 - 1).to(torch.bfloat16)
    w = (torch.rand(K, 3, 3, C // G, device=dev) * 0.1).to(torch.bfloat16)
    y = gpu_impl("conv2d_nhwc")(x, w, None, (1, 1), (1, 1), (1, 1), G)
    xr = x.float().permute(0, 3, 1, 2)
    wr = w.float().permute(0, 3, 1, 2)
    ref = torch.nn.functional.conv2d(xr, wr, None, 1, 1, 1, groups=G)
    ref = ref.permute(0, 2, 3, 1)
    err = (y.float() - ref).abs().max().item()
    scale = ref.abs().max().item()
    assert err < 0.02 * scale, (err, scale)


def test_fp8_grad_gemms_vs_fp32():
    """MX fp8 gradient GEMMs (e5m2 grads x e4m3 operands over transposed
    quantization) vs the fp32 oracle.  fp8 tolerances are statistical:
    require high correlation + bounded relative error."""
    torch.manual_seed(3)
    M, N, K = 512, 256, 384  # M,N %128, K %32
    dy = _rt((M, N)).to(BF)
    x = _rt((M, K)).to(BF)
    w = _rt((N, K), 0.3).to(BF)

    dy8, dys = _C.mx_cast(dy.contiguous(), True)
    wt8, wts = _C.mx_cast_t(w.contiguous(), False)
    dx = _C.gemm_mx(dy8, dys, wt8, wts, None, False, 1, 0)
    dx_ref = dy.float() @ w.float()
    corr = torch.corrcoef(torch.stack([dx.float().ravel(),
                                       dx_ref.ravel()]))[0, 1].item()
    assert corr > 0.995, f"dx decorrelated: {corr}"
    rel = (dx.float() - dx_ref).abs().max() / (dx_ref.abs().max() + 1e-6)
    assert rel < 0.15, f"dx rel err {rel}"

    dyt8, dyts = _C.mx_cast_t(dy.contiguous(), True)
    xt8, xts = _C.mx_cast_t(x.contiguous(), False)
    dw = _C.gemm_mx(dyt8, dyts, xt8, xts, None, False, 1, 0)
    dw_ref = dy.float().t() @ x.float()
    corr = torch.corrcoef(torch.stack([dw.float().ravel(),
                                       dw_ref.ravel()]))[0, 1].item()
    assert corr > 0.995, f"dw decorrelated: {corr}"
    rel = (dw.float() - dw_ref).abs().max() / (dw_ref.abs().max() + 1e-6)
    assert rel < 0.15, f"dw rel err {rel}"


def test_mx_cast_t_roundtrip():
    """Transposed MX cast: dequantized Q[c, r] must reconstruct x[r, c]
    within e4m3 block-quant error."""
    torch.manual_seed(4)
    R, C = 128, 100
    x = _rt((R, C)).to(BF)
    q, sc = _C.mx_cast_t(x.contiguous(), False)
    assert q.shape == (C, R) and sc.shape == (C, R // 32)
    # dequantize on CPU
    qf = q.float().cpu()  # not the value — need e4m3 decode via torch
    qv = q.cpu().numpy().view("uint8")
    import numpy as _np
    t = torch.from_numpy(qv.copy()).view(torch.float8_e4m3fn).float()
    scales = torch.pow(2.0, sc.cpu().float() - 127.0)
    t = t.reshape(C, R // 32, 32) * scales[:, :, None]
    back = t.reshape(C, R).t()
    ref = x.float().cpu()
    err = (back - ref).abs().max().item()
    amax = ref.abs().max().item()
    assert err / amax < 0.08, f"mx_cast_t roundtrip err {err} vs {amax}"


def test_gemm_i8_exact_vs_integer_oracle():
    """int8 MFMA GEMM accumulates int32 exactly — bit-identical to the
    integer oracle (the strongest kind of kernel test)."""
    torch.manual_seed(9)
    for (M, N, K) in [(128, 128, 128), (100, 200, 256), (512, 96, 384)]:
        a = torch.randint(-127, 128, (M, K), dtype=torch.int8,
                          device=_dev())
        b = torch.randint(-127, 128, (N, K), dtype=torch.int8,
                          device=_dev())
        out = _C.gemm_i8(a, b)
        ref = (a.cpu().to(torch.int32) @ b.cpu().to(torch.int32).t())
        assert (out.cpu() == ref).all(), f"i8 gemm mismatch {M}x{N}x{K}"
    # fused dequant epilogue
    a = torch.randint(-127, 128, (64, 128), dtype=torch.int8, device=_dev())
    b = torch.randint(-127, 128, (32, 128), dtype=torch.int8, device=_dev())
    isc = torch.tensor([1.0 / (127.0 * 127.0)], device=_dev())
    outf = _C.gemm_i8(a, b, isc)
    reff = (a.cpu().to(torch.int32) @ b.cpu().to(torch.int32).t()).float() \
        .to(_dev()) / (127.0 * 127.0)
    assert_close(outf, reff, rtol=1e-2, atol=1e-2, msg="i8 dequant")


def test_flash_cross_attention_fwd_bwd():
    """Tq != Tk (decoder enc-dec attention) through the fused flash path,
    forward AND backward vs the unfused fp32 oracle."""
    B, H, Tq, Tk, D = 2, 4, 128, 384, 64
    scale = 1.0 / (D ** 0.5)
    q = _rt((B, Tq, H * D)).to(BF).requires_grad_(True)
    k = _rt((B, Tk, H * D)).to(BF).requires_grad_(True)
    v = _rt((B, Tk, H * D)).to(BF).requires_grad_(True)

    out, m, l = _C.flash_fwd(q.detach().contiguous(),
                             k.detach().contiguous(),
                             v.detach().contiguous(), H, scale, False,
                             0.0, 0)
    qh = _heads(q.detach(), H).float()
    kh = _heads(k.detach(), H).float()
    vh = _heads(v.detach(), H).float()
    s = qh @ kh.transpose(-1, -2) * scale
    p = torch.nn.functional.softmax(s, dim=-1)
    ref = (p @ vh).transpose(1, 2).reshape(B, Tq, H * D)
    assert_close(out, ref, rtol=5e-2, atol=5e-2, msg="cross-attn out")

    dout = _rt((B, Tq, H * D)).to(BF)
    dq, dk, dv = _C.flash_bwd(q.detach().contiguous(),
                              k.detach().contiguous(),
                              v.detach().contiguous(), out.contiguous(),
                              dout.contiguous(), m, l, H, scale, False,
                              0.0, 0)
    # fp32 oracle grads
    qf = _heads(q.detach(), H).float().requires_grad_(True)
    kf = _heads(k.detach(), H).float().requires_grad_(True)
    vf = _heads(v.detach(), H).float().requires_grad_(True)
    sf = qf @ kf.transpose(-1, -2) * scale
    pf = torch.nn.functional.softmax(sf, dim=-1)
    of = (pf @ vf).transpose(1, 2).reshape(B, Tq, H * D)
    of.backward(dout.float())
    dq_ref = qf.grad.transpose(1, 2).reshape(B, Tq, H * D)
    dk_ref = kf.grad.transpose(1, 2).reshape(B, Tk, H * D)
    dv_ref = vf.grad.transpose(1, 2).reshape(B, Tk, H * D)
    assert_close(dq, dq_ref, rtol=6e-2, atol=6e-2, msg="cross dq")
    assert_close(dk, dk_ref, rtol=6e-2, atol=6e-2, msg="cross dk")
    assert_close(dv, dv_ref, rtol=6e-2, atol=6e-2, msg="cross dv")


@pytest.mark.parametrize("cfg", [
    (1, 2, 200, 200, True),    # causal, non-%128
    (2, 2, 100, 100, False),
    (1, 3, 160, 272, False),   # cross-attention with tails
    (2, 1, 64, 512, False),
])
def test_flash_tail_lengths(cfg):
    """Sequence lengths not % 128: tail blocks clamp staging and mask
    scores; fwd + bwd vs the fp32 oracle."""
    B, H, Tq, Tk, causal = cfg
    D = 64
    scale = 1.0 / (D ** 0.5)
    q = _rt((B, Tq, H * D)).to(BF)
    k = _rt((B, Tk, H * D)).to(BF)
    v = _rt((B, Tk, H * D)).to(BF)
    out, m, l = _C.flash_fwd(q.contiguous(), k.contiguous(),
                             v.contiguous(), H, scale, causal, 0.0, 0)
    qh, kh, vh = (_heads(x, H).float() for x in (q, k, v))
    s = qh @ kh.transpose(-1, -2) * scale
    if causal:
        cm = torch.triu(torch.ones(Tq, Tk, device=_dev(),
                                   dtype=torch.bool), 1)
        s = s.masked_fill(cm, float("-inf"))
    p = torch.nn.functional.softmax(s, dim=-1)
    ref = (p @ vh).transpose(1, 2).reshape(B, Tq, H * D)
    assert_close(out, ref, rtol=5e-2, atol=5e-2, msg=f"tail fwd {cfg}")

    dout = _rt((B, Tq, H * D)).to(BF)
    dq, dk, dv = _C.flash_bwd(q.contiguous(), k.contiguous(),
                              v.contiguous(), out.contiguous(),
                              dout.contiguous(), m, l, H, scale, causal,
                              0.0, 0)
    qf = qh.clone().requires_grad_(True)
    kf = kh.clone().requires_grad_(True)
    vf = vh.clone().requires_grad_(True)
    sf = qf @ kf.transpose(-1, -2) * scale
    if causal:
        sf = sf.masked_fill(cm, float("-inf"))
    pf = torch.nn.functional.softmax(sf, dim=-1)
    (pf @ vf).transpose(1, 2).reshape(B, Tq, H * D).backward(dout.float())
    assert_close(dq, qf.grad.transpose(1, 2).reshape(B, Tq, H * D),
                 rtol=7e-2, atol=7e-2, msg=f"tail dq {cfg}")
    assert_close(dk, kf.grad.transpose(1, 2).reshape(B, Tk, H * D),
                 rtol=7e-2, atol=7e-2, msg=f"tail dk {cfg}")
    assert_close(dv, vf.grad.transpose(1, 2).reshape(B, Tk, H * D),
                 rtol=7e-2, atol=7e-2, msg=f"tail dv {cfg}")


def test_ew_unary_family_vs_torch():
    """Generic elementwise engine (ew.hip): every unary functor fwd+bwd
    vs torch fp32 (reference mxnet_op.h Kernel::Launch surface)."""
    from anand_mxnet_amd.ops.hip import EW_UNARY, _EwUnary
    torch.manual_seed(1)
    refs = {
        "relu": torch.relu, "sigmoid": torch.sigmoid, "tanh": torch.tanh,
        "exp": torch.exp, "log": torch.log, "sqrt": torch.sqrt,
        "rsqrt": torch.rsqrt, "square": torch.square, "abs": torch.abs,
        "negative": torch.neg, "reciprocal": torch.reciprocal,
        "sin": torch.sin, "cos": torch.cos, "erf": torch.erf,
        "log1p": torch.log1p, "expm1": torch.expm1, "floor": torch.floor,
        "ceil": torch.ceil, "round": torch.round, "sign": torch.sign,
        "softsign": lambda t: t / (1 + t.abs()),
    }
    for name, opid in EW_UNARY.items():
        pos_only = name in ("log", "sqrt", "rsqrt", "log1p", "reciprocal")
        x = torch.rand(1000, device=_dev()) * 2 + 0.1 if pos_only else \
            (torch.rand(1000, device=_dev()) * 4 - 2)
        xb = x.to(BF)
        y = _C.ew_unary(opid, xb.contiguous())
        ref = refs[name](xb.float())
        assert_close(y, ref, rtol=2e-2, atol=2e-2, msg=f"ew {name}")
        # gradient (skip zero-grad step functions)
        if name in ("floor", "ceil", "round", "sign"):
            continue
        xg = xb.clone().requires_grad_(True)
        yy = _EwUnary.apply(xg, opid)
        dy = torch.rand_like(yy)
        yy.backward(dy)
        xr = xb.float().clone().requires_grad_(True)
        refs[name](xr).backward(dy.float())
        assert_close(xg.grad, xr.grad, rtol=4e-2, atol=4e-2,
                     msg=f"ew grad {name}")


def test_ew_binary_and_broadcast():
    from anand_mxnet_amd.ops.hip import _EwBinary
    torch.manual_seed(2)
    a = (torch.rand(64, 33, device=_dev()) * 2 - 1).to(BF)
    b = (torch.rand(64, 33, device=_dev()) * 2 + 0.5).to(BF)
    for opid, fn in [(0, torch.add), (1, torch.sub), (2, torch.mul),
                     (3, torch.div), (5, torch.maximum),
                     (6, torch.minimum)]:
        y = _C.ew_binary(opid, a.contiguous(), b.contiguous())
        assert_close(y, fn(a.float(), b.float()), rtol=2e-2, atol=2e-2,
                     msg=f"ew bin {opid}")
        # autograd path
        ag = a.clone().requires_grad_(True)
        bg = b.clone().requires_grad_(True)
        yy = _EwBinary.apply(ag, bg, opid)
        dy = torch.rand_like(yy)
        yy.backward(dy)
        ar = a.float().clone().requires_grad_(True)
        br = b.float().clone().requires_grad_(True)
        fn(ar, br).backward(dy.float())
        assert_close(ag.grad, ar.grad, rtol=4e-2, atol=4e-2,
                     msg=f"ew bin da {opid}")
        assert_close(bg.grad, br.grad, rtol=4e-2, atol=4e-2,
                     msg=f"ew bin db {opid}")
    # broadcast forward (inference)
    with torch.no_grad():
        c = (torch.rand(1, 33, device=_dev()) * 2 - 1).to(BF)
        y = _C.ew_binary(2, a.contiguous(), c.contiguous())
        assert_close(y, a.float() * c.float(), rtol=2e-2, atol=2e-2,
                     msg="ew bcast mul")
        d = (torch.rand(64, 1, device=_dev())).to(BF)
        y = _C.ew_binary(0, a.contiguous(), d.contiguous())
        assert_close(y, a.float() + d.float(), rtol=2e-2, atol=2e-2,
                     msg="ew bcast add")


def test_ew_reduce_last():
    torch.manual_seed(3)
    x = (torch.rand(37, 515, device=_dev()) * 2 - 1).to(BF)
    s = _C.ew_reduce_last(0, x.contiguous())
    assert_close(s, x.float().sum(-1), rtol=2e-2, atol=5e-2, msg="ew sum")
    m = _C.ew_reduce_last(1, x.contiguous())
    assert_close(m, x.float().max(-1).values, rtol=1e-4, atol=1e-4,
                 msg="ew max")
    mn = _C.ew_reduce_last(2, x.contiguous())
    assert_close(mn, x.float().min(-1).values, rtol=1e-4, atol=1e-4,
                 msg="ew min")


def test_native_sampling_moments():
    """Counter-hash sampling kernels: statistical moments (reference
    test_random.py pattern)."""
    from anand_mxnet_amd.base import lib
    C = lib()
    n = 1 << 20
    u = C.rng_sample([n], -1.0, 3.0, 12345, False, 0).float()
    assert abs(u.mean().item() - 1.0) < 0.02
    assert abs(u.var().item() - 16.0 / 12.0) < 0.03
    assert u.min().item() >= -1.0 and u.max().item() <= 3.0
    g = C.rng_sample([n], 0.5, 2.0, 999, True, 0).float()
    assert abs(g.mean().item() - 0.5) < 0.03
    assert abs(g.std().item() - 2.0) < 0.05
    # skew ~ 0, kurtosis ~ 3 (loose bf16 bounds)
    z = (g - g.mean()) / g.std()
    assert abs((z ** 3).mean().item()) < 0.05
    assert abs((z ** 4).mean().item() - 3.0) < 0.2
    # different salts decorrelate
    h = C.rng_sample([n], 0.5, 2.0, 1000, True, 0).float()
    corr = torch.corrcoef(torch.stack([g, h]))[0, 1].item()
    assert abs(corr) < 0.01


def test_take_via_embedding_gather():
    w = _rt((128, 64)).to(BF).requires_grad_(True)
    idx = torch.randint(0, 128, (10, 7), device=_dev())
    from anand_mxnet_amd.ops.registry import get as get_op
    import anand_mxnet_amd  # noqa
    from anand_mxnet_amd import nd
    out = get_op("take")(nd.NDArray(w), nd.NDArray(idx), axis=0)
    o = out.asnumpy()
    ref = w.detach().float()[idx.reshape(-1)].reshape(10, 7, 64)
    np.testing.assert_allclose(o.astype(np.float32),
                               ref.cpu().numpy(), rtol=1e-2, atol=1e-2)


@pytest.mark.parametrize("cfg", [
    (2, 2, 256, 256, False),
    (1, 4, 128, 384, False),   # cross-attention
    (1, 2, 200, 200, True),    # causal, tails
])
def test_flash_fwd_d128(cfg):
    """D = 128 heads: 64-chunk flash forward + 32-q-chunk backward."""
    B, H, Tq, Tk, causal = cfg
    D = 128
    scale = 1.0 / (D ** 0.5)
    q = _rt((B, Tq, H * D)).to(BF)
    k = _rt((B, Tk, H * D)).to(BF)
    v = _rt((B, Tk, H * D)).to(BF)
    out, m, l = _C.flash_fwd(q.contiguous(), k.contiguous(),
                             v.contiguous(), H, scale, causal, 0.0, 0)

    def heads(x, T):
        return x.reshape(B, T, H, D).transpose(1, 2).float()
    s = heads(q, Tq) @ heads(k, Tk).transpose(-1, -2) * scale
    if causal:
        cm = torch.triu(torch.ones(Tq, Tk, device=_dev(),
                                   dtype=torch.bool), 1)
        s = s.masked_fill(cm, float("-inf"))
    p = torch.nn.functional.softmax(s, dim=-1)
    ref = (p @ heads(v, Tk)).transpose(1, 2).reshape(B, Tq, H * D)
    assert_close(out, ref, rtol=5e-2, atol=5e-2, msg=f"d128 fwd {cfg}")
    ref_m = s.max(-1).values.reshape(B * H, Tq)
    assert_close(m, ref_m, rtol=1e-2, atol=1e-2, msg="d128 m")

    dout = _rt((B, Tq, H * D)).to(BF)
    dq, dk, dv = _C.flash_bwd(q.contiguous(), k.contiguous(),
                              v.contiguous(), out.contiguous(),
                              dout.contiguous(), m, l, H, scale, causal,
                              0.0, 0)
    qf = heads(q, Tq).clone().requires_grad_(True)
    kf = heads(k, Tk).clone().requires_grad_(True)
    vf = heads(v, Tk).clone().requires_grad_(True)
    sf = qf @ kf.transpose(-1, -2) * scale
    if causal:
        sf = sf.masked_fill(cm, float("-inf"))
    pf = torch.nn.functional.softmax(sf, dim=-1)
    (pf @ vf).transpose(1, 2).reshape(B, Tq, H * D).backward(dout.float())
    assert_close(dq, qf.grad.transpose(1, 2).reshape(B, Tq, H * D),
                 rtol=7e-2, atol=7e-2, msg=f"d128 dq {cfg}")
    assert_close(dk, kf.grad.transpose(1, 2).reshape(B, Tk, H * D),
                 rtol=7e-2, atol=7e-2, msg=f"d128 dk {cfg}")
    assert_close(dv, vf.grad.transpose(1, 2).reshape(B, Tk, H * D),
                 rtol=7e-2, atol=7e-2, msg=f"d128 dv {cfg}")


def test_gemm_wg_bias_fused():
    """gemm_wg_bias: dW and the folded bias gradient (colsum of A) in one
    pass vs fp32 oracle, with and without split-K slots."""
    torch.manual_seed(7)
    for (m, Mo, No) in [(4096, 768, 768), (100352, 128, 128),
                        (8192, 3072, 768), (777, 64, 64)]:
        dy = _rt((m, Mo)).to(BF)
        x = _rt((m, No)).to(BF)
        dw, db = _C.gemm_wg_bias(dy.contiguous(), x.contiguous())
        ref_dw = dy.float().t() @ x.float()
        ref_db = dy.float().sum(0)
        assert_close(dw, ref_dw, rtol=5e-2, atol=5e-2,
                     msg=f"wg_bias dw {m}x{Mo}x{No}")
        assert_close(db, ref_db, rtol=2e-2, atol=2e-2,
                     msg=f"wg_bias db {m}x{Mo}x{No}")


def test_fold_batchnorm_gpu_bf16():
    """fold_batchnorm on a bf16 NHWC conv-bn-relu stack on the GPU:
    folded (fused conv relu epilogue) matches unfolded inference."""
    import numpy as np
    import anand_mxnet_amd as mx
    from anand_mxnet_amd.gluon import nn
    from anand_mxnet_amd.gluon.utils import fold_batchnorm
    net = nn.HybridSequential()
    with net.name_scope():
        net.add(nn.Conv2D(64, 3, padding=1, layout="NHWC", in_channels=64,
                          use_bias=False))
        net.add(nn.BatchNorm(axis=3, fuse_relu=True, in_channels=64))
        net.add(nn.Conv2D(128, 1, layout="NHWC", in_channels=64))
        net.add(nn.BatchNorm(axis=3, in_channels=128))
    net.cast("bfloat16")
    net.initialize(mx.init.Xavier(), ctx=mx.gpu(0))
    x = mx.nd.random.uniform(-1, 1, shape=(2, 16, 16, 64),
                             dtype="bfloat16", ctx=mx.gpu(0))
    net(x)  # materialize
    rs = np.random.RandomState(0)
    for blk in net._children.values():
        if isinstance(blk, nn.BatchNorm):
            n = blk.running_mean.shape[0]
            blk.running_mean.set_data(mx.nd.array(
                (rs.rand(n) - 0.5).astype("float32"), ctx=mx.gpu(0)))
            blk.running_var.set_data(mx.nd.array(
                (0.5 + rs.rand(n)).astype("float32"), ctx=mx.gpu(0)))
    ref = net(x).asnumpy().astype("float32")
    assert fold_batchnorm(net) == 2
    out = net(x).asnumpy().astype("float32")
    assert np.allclose(out, ref, rtol=5e-2, atol=5e-2), \
        np.abs(out - ref).max()


def test_mha_d128_op():
    """nd.mha with 128-dim heads dispatches the fused flash path
    (forward+backward) and matches the unfused fp32 oracle."""
    B, T, H, D = 2, 200, 2, 128
    q = _rt((B, T, H * D)).to(BF).requires_grad_(True)
    k = _rt((B, T, H * D)).to(BF).requires_grad_(True)
    v = _rt((B, T, H * D)).to(BF).requires_grad_(True)
    from anand_mxnet_amd.ops.hip import mha
    scale = 1.0 / (D ** 0.5)
    out = mha(q, k, v, H, True, scale)
    dout = _rt((B, T, H * D)).to(BF)
    out.backward(dout)
    qf = _heads(q.detach(), H).float().requires_grad_(True)
    kf = _heads(k.detach(), H).float().requires_grad_(True)
    vf = _heads(v.detach(), H).float().requires_grad_(True)
    s = qf @ kf.transpose(-1, -2) * scale
    cm = torch.triu(torch.ones(T, T, device=_dev(), dtype=torch.bool), 1)
    s = s.masked_fill(cm, float("-inf"))
    p = torch.nn.functional.softmax(s, dim=-1)
    (p @ vf).transpose(1, 2).reshape(B, T, H * D).backward(dout.float())
    assert_close(out, (p @ vf).transpose(1, 2).reshape(B, T, H * D),
                 rtol=5e-2, atol=5e-2, msg="mha d128 out")
    assert_close(q.grad, qf.grad.transpose(1, 2).reshape(B, T, H * D),
                 rtol=7e-2, atol=7e-2, msg="mha d128 dq")
    assert_close(k.grad, kf.grad.transpose(1, 2).reshape(B, T, H * D),
                 rtol=7e-2, atol=7e-2, msg="mha d128 dk")
    assert_close(v.grad, vf.grad.transpose(1, 2).reshape(B, T, H * D),
                 rtol=7e-2, atol=7e-2, msg="mha d128 dv")


def test_window_gemm_routes_vs_oracle():
    """Shapes large enough to take the 2-barrier-window kernels
    (gemm_tn8g_core: >=240 256^2-tiles & K>=1024; gemm_nn8g_core:
    >=160 tiles & K>=1024) match the fp32 oracle."""
    torch.manual_seed(11)
    # TN route: 4096x4096, K=1024 -> 256 tiles
    a = _rt((4096, 1024)).to(BF)
    b = _rt((4096, 1024)).to(BF)
    y = _C.gemm_tn(a.contiguous(), b.contiguous(), None, False)
    ref = a.float() @ b.float().t()
    assert_close(y, ref, rtol=3e-2, atol=3e-2, msg="tn8g route")
    # NN route: 4096x2560, K=1024 -> 160 tiles
    a = _rt((4096, 1024)).to(BF)
    b = _rt((1024, 2560)).to(BF)
    y = _C.gemm_nn(a.contiguous(), b.contiguous())
    ref = a.float() @ b.float()
    assert_close(y, ref, rtol=3e-2, atol=3e-2, msg="nn8g route")
    # tail rows on both routes (M not % 256)
    a = _rt((4000, 1024)).to(BF)
    b = _rt((4096, 1024)).to(BF)
    y = _C.gemm_tn(a.contiguous(), b.contiguous(), None, False)
    assert_close(y, a.float() @ b.float().t(), rtol=3e-2, atol=3e-2,
                 msg="tn8g tail")
