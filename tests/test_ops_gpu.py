"""GPU numerics tests: every HIP kernel vs a plain PyTorch fp32 reference
(SURVEY.md §4 strategy). All marked @pytest.mark.gpu."""

import os

import pytest
import torch

pytestmark = pytest.mark.gpu


def _cl(x):
    return x.contiguous(memory_format=torch.channels_last)


@pytest.fixture(scope="module")
def ext():
    from deepfake_detection_amd.ops.extension import load_extension

    return load_extension()


# ---------------------------------------------------------------------------
# normalize
# ---------------------------------------------------------------------------
@pytest.mark.parametrize("dtype", ["float32", "bfloat16", "float16"])
def test_normalize_uint8(ext, dtype):
    torch.manual_seed(0)
    x = torch.randint(0, 256, (3, 12, 37, 41), dtype=torch.uint8, device="cuda")
    mean = torch.rand(12, device="cuda") * 255
    std = torch.rand(12, device="cuda") * 100 + 20
    out = ext.normalize_uint8_nhwc(x, mean, std, dtype, True)
    ref = (x.float() - mean.view(1, 12, 1, 1)) / std.view(1, 12, 1, 1)
    assert out.is_contiguous(memory_format=torch.channels_last)
    tol = 1e-5 if dtype == "float32" else (0.02 if dtype == "bfloat16" else 0.005)
    assert torch.allclose(out.float(), ref, atol=tol, rtol=tol)


# ---------------------------------------------------------------------------
# fused BN + act
# ---------------------------------------------------------------------------
@pytest.mark.parametrize("training", [True, False])
@pytest.mark.parametrize("act", ["silu", "relu", "none"])
@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_bn_act_forward(ext, training, act, dtype):
    torch.manual_seed(1)
    N, C, H, W = 4, 48, 17, 19
    x = _cl(torch.randn(N, C, H, W, device="cuda", dtype=dtype))
    weight = torch.randn(C, device="cuda") * 0.5 + 1
    bias = torch.randn(C, device="cuda") * 0.1
    rm = torch.randn(C, device="cuda") * 0.1
    rv = torch.rand(C, device="cuda") + 0.5
    rm_ref, rv_ref = rm.clone(), rv.clone()
    momentum, eps = 0.01, 1e-3

    y, mean, invstd = ext.bn_act_fwd(x, weight, bias, rm, rv, training, momentum, eps, act)

    ref = torch.nn.functional.batch_norm(
        x.float(), rm_ref, rv_ref, weight, bias, training, momentum, eps)
    if act == "silu":
        ref = torch.nn.functional.silu(ref)
    elif act == "relu":
        ref = torch.nn.functional.relu(ref)

    tol = 1e-4 if dtype == torch.float32 else 0.05
    assert torch.allclose(y.float(), ref, atol=tol, rtol=tol)
    # running stats parity (fp32 path, exact math)
    assert torch.allclose(rm, rm_ref, atol=1e-4, rtol=1e-4)
    assert torch.allclose(rv, rv_ref, atol=1e-3, rtol=1e-3)


@pytest.mark.parametrize("training", [True, False])
@pytest.mark.parametrize("act", ["silu", "none"])
def test_bn_act_backward(ext, training, act):
    torch.manual_seed(2)
    N, C, H, W = 3, 32, 13, 11
    x = _cl(torch.randn(N, C, H, W, device="cuda"))
    weight = torch.randn(C, device="cuda") * 0.5 + 1
    bias = torch.randn(C, device="cuda") * 0.1
    rm = torch.zeros(C, device="cuda")
    rv = torch.ones(C, device="cuda")
    dy = _cl(torch.randn(N, C, H, W, device="cuda"))
    eps = 1e-3

    y, mean, invstd = ext.bn_act_fwd(x, weight, bias, rm.clone(), rv.clone(),
                                     training, 0.1, eps, act)
    dx, dgamma, dbeta = ext.bn_act_bwd(dy, x, weight, bias, mean, invstd, training, act)

    # torch autograd reference in fp32
    x_ref = x.float().detach().requires_grad_(True)
    w_ref = weight.detach().requires_grad_(True)
    b_ref = bias.detach().requires_grad_(True)
    ref = torch.nn.functional.batch_norm(
        x_ref, rm.clone(), rv.clone(), w_ref, b_ref, training, 0.1, eps)
    if act == "silu":
        ref = torch.nn.functional.silu(ref)
    ref.backward(dy.float())

    assert torch.allclose(y.float(), ref.detach(), atol=1e-4, rtol=1e-4)
    assert torch.allclose(dx.float(), x_ref.grad, atol=1e-3, rtol=1e-3)
    assert torch.allclose(dgamma, w_ref.grad, atol=1e-2, rtol=1e-3)
    assert torch.allclose(dbeta, b_ref.grad, atol=1e-2, rtol=1e-3)


def test_bn_act_autograd_function():
    from deepfake_detection_amd.ops.bn_act import fused_bn_act

    torch.manual_seed(3)
    N, C, H, W = 2, 24, 9, 9
    x = _cl(torch.randn(N, C, H, W, device="cuda")).requires_grad_(True)
    bn = torch.nn.BatchNorm2d(C, momentum=0.01, eps=1e-3).cuda()
    y = fused_bn_act(x, bn.weight, bn.bias, bn.running_mean, bn.running_var,
                     True, 0.01, 1e-3, "silu")
    loss = (y ** 2).sum()
    loss.backward()
    assert x.grad is not None and torch.isfinite(x.grad).all()
    assert bn.weight.grad is not None and torch.isfinite(bn.weight.grad).all()


# ---------------------------------------------------------------------------
# SE
# ---------------------------------------------------------------------------
@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_se_forward_backward(ext, dtype):
    from deepfake_detection_amd.ops.se import fused_se

    torch.manual_seed(4)
    N, C, H, W, Cr = 3, 64, 11, 13, 16
    x0 = torch.randn(N, C, H, W, device="cuda", dtype=dtype)
    w1 = torch.randn(Cr, C, 1, 1, device="cuda") * 0.1
    b1 = torch.randn(Cr, device="cuda") * 0.1
    w2 = torch.randn(C, Cr, 1, 1, device="cuda") * 0.1
    b2 = torch.randn(C, device="cuda") * 0.1

    x = _cl(x0.clone()).requires_grad_(True)
    w1_p = w1.clone().requires_grad_(True)
    b1_p = b1.clone().requires_grad_(True)
    w2_p = w2.clone().requires_grad_(True)
    b2_p = b2.clone().requires_grad_(True)
    y = fused_se(x, w1_p, b1_p, w2_p, b2_p, "silu")
    dy = torch.randn_like(y)
    y.backward(dy)

    # fp32 torch reference
    xr = x0.float().detach().requires_grad_(True)
    w1r = w1.detach().requires_grad_(True)
    b1r = b1.detach().requires_grad_(True)
    w2r = w2.detach().requires_grad_(True)
    b2r = b2.detach().requires_grad_(True)
    s = xr.mean(dim=(2, 3), keepdim=True)
    t = torch.nn.functional.conv2d(s, w1r, b1r)
    t = torch.nn.functional.silu(t)
    t = torch.nn.functional.conv2d(t, w2r, b2r)
    yr = xr * torch.sigmoid(t)
    yr.backward(dy.float())

    tol = 1e-4 if dtype == torch.float32 else 0.05
    assert torch.allclose(y.float(), yr.detach(), atol=tol, rtol=tol)
    assert torch.allclose(x.grad.float(), xr.grad, atol=tol * 10, rtol=tol * 10)
    assert torch.allclose(w1_p.grad.float(), w1r.grad, atol=0.05, rtol=0.02)
    assert torch.allclose(w2_p.grad.float(), w2r.grad, atol=0.05, rtol=0.02)
    assert torch.allclose(b1_p.grad.float(), b1r.grad, atol=0.05, rtol=0.02)
    assert torch.allclose(b2_p.grad.float(), b2r.grad, atol=0.05, rtol=0.02)


# ---------------------------------------------------------------------------
# global avg pool
# ---------------------------------------------------------------------------
def test_global_avg_pool(ext):
    from deepfake_detection_amd.ops.pool import fused_global_avg_pool

    torch.manual_seed(5)
    x = _cl(torch.randn(4, 96, 19, 19, device="cuda")).requires_grad_(True)
    y = fused_global_avg_pool(x)
    assert y.shape == (4, 96)
    ref = x.float().mean(dim=(2, 3))
    assert torch.allclose(y.float(), ref, atol=1e-5, rtol=1e-5)
    dy = torch.randn_like(y)
    y.backward(dy)
    xr = x.detach().clone().requires_grad_(True)
    xr.mean(dim=(2, 3)).backward(dy)
    assert torch.allclose(x.grad, xr.grad, atol=1e-6)


# ---------------------------------------------------------------------------
# fused optimizers
# ---------------------------------------------------------------------------
def test_rmsproptf_fused_matches_cpu():
    from deepfake_detection_amd.optim import RMSpropTF

    torch.manual_seed(6)
    shapes = [(37,), (16, 8), (128,), (5, 5, 3, 3)]
    cpu_params = [torch.nn.Parameter(torch.randn(s)) for s in shapes]
    gpu_params = [torch.nn.Parameter(p.detach().clone().cuda()) for p in cpu_params]
    kw = dict(lr=0.05, alpha=0.9, eps=1e-3, momentum=0.9, weight_decay=1e-4)
    opt_cpu = RMSpropTF(cpu_params, **kw)
    opt_gpu = RMSpropTF(gpu_params, **kw)
    for step in range(5):
        torch.manual_seed(100 + step)
        grads = [torch.randn(s) for s in shapes]
        for p, g in zip(cpu_params, grads):
            p.grad = g.clone()
        for p, g in zip(gpu_params, grads):
            p.grad = g.clone().cuda()
        opt_cpu.step()
        opt_gpu.step()
    for pc, pg in zip(cpu_params, gpu_params):
        assert torch.allclose(pc.detach(), pg.detach().cpu(), atol=1e-5, rtol=1e-5)


def test_adamw_fused_matches_cpu():
    from deepfake_detection_amd.optim import AdamW

    torch.manual_seed(7)
    shapes = [(41,), (8, 8)]
    cpu_params = [torch.nn.Parameter(torch.randn(s)) for s in shapes]
    gpu_params = [torch.nn.Parameter(p.detach().clone().cuda()) for p in cpu_params]
    kw = dict(lr=0.01, weight_decay=0.1)
    opt_cpu = AdamW(cpu_params, **kw)
    opt_gpu = AdamW(gpu_params, **kw)
    for step in range(4):
        torch.manual_seed(200 + step)
        grads = [torch.randn(s) for s in shapes]
        for p, g in zip(cpu_params, grads):
            p.grad = g.clone()
        for p, g in zip(gpu_params, grads):
            p.grad = g.clone().cuda()
        opt_cpu.step()
        opt_gpu.step()
    for pc, pg in zip(cpu_params, gpu_params):
        assert torch.allclose(pc.detach(), pg.detach().cpu(), atol=1e-5, rtol=1e-5)


def test_ema_fused(ext):
    e = [torch.ones(100, device="cuda"), torch.full((31,), 2.0, device="cuda")]
    m = [torch.full((100,), 2.0, device="cuda"), torch.full((31,), 4.0, device="cuda")]
    ext.ema_multi_tensor(e, m, 0.9)
    assert torch.allclose(e[0], torch.full((100,), 1.1, device="cuda"))
    assert torch.allclose(e[1], torch.full((31,), 2.2, device="cuda"))


# ---------------------------------------------------------------------------
# end-to-end model A/B: fused HIP path vs plain torch ops on the same device
# ---------------------------------------------------------------------------
def test_model_fused_vs_torch_path():
    import deepfake_detection_amd as dfd

    torch.manual_seed(8)
    model = dfd.create_model("efficientnet_b0", num_classes=2).cuda().eval()
    model = model.to(memory_format=torch.channels_last)
    x = torch.randn(2, 3, 224, 224, device="cuda").contiguous(
        memory_format=torch.channels_last)

    with torch.no_grad():
        y_fused = model(x)
        os.environ["DFD_AMD_FORCE_TORCH_OPS"] = "1"
        try:
            y_torch = model(x)
        finally:
            del os.environ["DFD_AMD_FORCE_TORCH_OPS"]
    assert torch.allclose(y_fused, y_torch, atol=1e-3, rtol=1e-3)


def test_train_step_gpu():
    """One forward+backward+step of the production model family on GPU,
    bf16 autocast, fused ops + fused optimizer."""
    import deepfake_detection_amd as dfd
    from deepfake_detection_amd.optim import RMSpropTF

    torch.manual_seed(9)
    model = dfd.create_model("efficientnet_b0", num_classes=2, in_chans=12).cuda()
    model = model.to(memory_format=torch.channels_last)
    opt = RMSpropTF(model.parameters(), lr=1e-4, alpha=0.9, eps=1e-3, momentum=0.9)
    x = torch.randn(4, 12, 64, 64, device="cuda").contiguous(
        memory_format=torch.channels_last)
    t = torch.randint(0, 2, (4,), device="cuda")
    with torch.autocast("cuda", torch.bfloat16):
        out = model(x)
        loss = torch.nn.functional.cross_entropy(out, t)
    loss.backward()
    opt.step()
    torch.cuda.synchronize()
    assert torch.isfinite(loss)


# ---------------------------------------------------------------------------
# depthwise conv (fwd / bwd-data / bwd-weight) vs fp32 torch conv2d
# ---------------------------------------------------------------------------
@pytest.mark.parametrize("k,stride", [(3, 1), (3, 2), (5, 1), (5, 2)])
@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_dwconv_matches_torch(k, stride, dtype):
    from deepfake_detection_amd.ops.dwconv import dw_conv2d

    torch.manual_seed(7)
    N, C, H, W = 3, 40, 23, 29  # C not a multiple of 16 to hit vec fallbacks
    pad = (k - 1) // 2
    x = _cl(torch.randn(N, C, H, W, device="cuda", dtype=dtype)).requires_grad_(True)
    w = torch.randn(C, 1, k, k, device="cuda", dtype=dtype, requires_grad=True)

    y = dw_conv2d(x, w, None, stride, pad)
    ref_x = x.detach().float().requires_grad_(True)
    ref_w = w.detach().float().requires_grad_(True)
    ref = torch.nn.functional.conv2d(ref_x, ref_w, None, stride, pad, 1, groups=C)

    tol = 1e-4 if dtype == torch.float32 else 0.08
    assert y.shape == ref.shape
    assert torch.allclose(y.float(), ref, atol=tol, rtol=tol)

    dy = torch.randn_like(ref)
    ref.backward(dy)
    y.backward(dy.to(dtype))
    assert torch.allclose(x.grad.float(), ref_x.grad, atol=tol * 5, rtol=tol * 5)
    # bwd-weight reduces over N*Ho*Wo values; scale tolerance accordingly
    wtol = 1e-3 if dtype == torch.float32 else 0.3
    assert torch.allclose(w.grad.float(), ref_w.grad, atol=wtol, rtol=0.05)


@pytest.mark.parametrize("vec_c", [8, 64, 144])
def test_dwconv_vector_widths(vec_c):
    from deepfake_detection_amd.ops.dwconv import dw_conv2d

    torch.manual_seed(8)
    x = _cl(torch.randn(2, vec_c, 15, 17, device="cuda"))
    w = torch.randn(vec_c, 1, 3, 3, device="cuda")
    y = dw_conv2d(x, w, None, 1, 1)
    ref = torch.nn.functional.conv2d(x, w, None, 1, 1, 1, groups=vec_c)
    assert torch.allclose(y, ref, atol=1e-4, rtol=1e-4)


def test_depthwise_module_routes_to_hip():
    """DepthwiseConv2d must run the HIP kernel on GPU (no silent MIOpen path)."""
    from deepfake_detection_amd.models.layers import DepthwiseConv2d, create_conv2d

    m = create_conv2d(32, 32, 3, stride=1, padding="", depthwise=True).cuda()
    assert isinstance(m, DepthwiseConv2d)
    x = _cl(torch.randn(2, 32, 19, 19, device="cuda"))
    y = m(x)
    ref = torch.nn.functional.conv2d(x, m.weight, m.bias, m.stride, m.padding,
                                     m.dilation, m.groups)
    assert torch.allclose(y, ref, atol=1e-4, rtol=1e-4)


# ---------------------------------------------------------------------------
# experimental MFMA pointwise conv vs torch fp32 conv
# ---------------------------------------------------------------------------
@pytest.mark.parametrize("m,k,n", [(4 * 19 * 19, 160, 960), (2 * 75 * 75, 192, 32),
                                   (3 * 38 * 38, 288, 48), (129, 24, 17)])
def test_pwconv_mfma_matches_torch(m, k, n):
    from deepfake_detection_amd.ops.pwconv import pw_conv2d_fwd

    torch.manual_seed(11)
    # express M as (B, H, W) = (1, 1, m) — the kernel only sees M = B*H*W
    x = torch.randn(1, k, 1, m, device="cuda", dtype=torch.bfloat16)
    x = x.contiguous(memory_format=torch.channels_last)
    w = torch.randn(n, k, 1, 1, device="cuda", dtype=torch.bfloat16)
    y = pw_conv2d_fwd(x, w)
    ref = torch.nn.functional.conv2d(x.float(), w.float())
    assert y.shape == ref.shape
    assert torch.allclose(y.float(), ref, atol=0.1 + 0.02 * (k ** 0.5), rtol=0.05)


@pytest.mark.parametrize("act", ["none", "silu"])
def test_bn_act_fused_residual(act):
    """y = act(bn(x)) + res in one kernel; residual grad = upstream grad."""
    from deepfake_detection_amd.ops.bn_act import fused_bn_act

    torch.manual_seed(3)
    N, C, H, W = 4, 48, 17, 19
    x = _cl(torch.randn(N, C, H, W, device="cuda")).requires_grad_(True)
    res = _cl(torch.randn(N, C, H, W, device="cuda")).requires_grad_(True)
    bn = torch.nn.BatchNorm2d(C, momentum=0.01, eps=1e-3).cuda()

    y = fused_bn_act(x, bn.weight, bn.bias, bn.running_mean, bn.running_var,
                     True, 0.01, 1e-3, act, res)
    rm = bn.running_mean.clone().zero_()
    rv = bn.running_var.clone().fill_(1)
    ref = torch.nn.functional.batch_norm(
        x.detach(), rm, rv, bn.weight.detach(), bn.bias.detach(), True, 0.01, 1e-3)
    if act == "silu":
        ref = torch.nn.functional.silu(ref)
    ref = ref + res.detach()
    assert torch.allclose(y, ref, atol=2e-4, rtol=2e-4)

    dy = torch.randn_like(y)
    y.backward(dy)
    assert torch.allclose(res.grad, dy)
    assert x.grad is not None and torch.isfinite(x.grad).all()


def test_pwconv_mfma_autograd():
    """fwd + bwd-data (same kernel, transposed weight) + bwd-weight (split-M
    MFMA kernel) vs fp32 torch conv."""
    from deepfake_detection_amd.ops.pwconv import pw_conv2d

    torch.manual_seed(12)
    B, K, N, H = 3, 96, 160, 13
    x = _cl(torch.randn(B, K, H, H, device="cuda", dtype=torch.bfloat16)).requires_grad_(True)
    w = torch.randn(N, K, 1, 1, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    y = pw_conv2d(x, w)
    rx = x.detach().float().requires_grad_(True)
    rw = w.detach().float().requires_grad_(True)
    ref = torch.nn.functional.conv2d(rx, rw)
    assert torch.allclose(y.float(), ref, atol=0.3, rtol=0.05)
    dy = torch.randn_like(ref)
    ref.backward(dy)
    y.backward(dy.to(torch.bfloat16))
    assert torch.allclose(x.grad.float(), rx.grad, atol=0.5, rtol=0.05)
    assert torch.allclose(w.grad.float(), rw.grad, atol=2.0, rtol=0.05)


def test_conv2d_same_depthwise_routes_to_hip():
    """tf_-model SAME-pad depthwise path: pad_same (torch) + HIP dw kernel
    with padding 0 must equal F.conv2d over the padded input."""
    from deepfake_detection_amd.models.layers import Conv2dSame

    torch.manual_seed(5)
    m = Conv2dSame(32, 32, 5, stride=2, groups=32, bias=False).cuda()
    x = _cl(torch.randn(2, 32, 27, 31, device="cuda"))
    y = m(x)
    from deepfake_detection_amd.models.layers import pad_same

    ref = torch.nn.functional.conv2d(
        pad_same(x, [5, 5], [2, 2]), m.weight, None, 2, 0, 1, 32)
    assert y.shape == ref.shape
    assert torch.allclose(y, ref, atol=1e-4, rtol=1e-4)


# ---------------------------------------------------------------------------
# production pointwise conv path: bwd-weight kernel, stats epilogue, routing
# ---------------------------------------------------------------------------
@pytest.mark.parametrize("m,n,k", [(2 * 75 * 75, 144, 24), (4 * 19 * 19, 448, 1792),
                                   (3 * 38 * 38, 48, 288), (2 * 10 * 10 + 3, 40, 80)])
def test_pwconv_wgrad_kernel_matches_torch(m, n, k):
    from deepfake_detection_amd.ops.extension import load_extension

    ext = load_extension()
    torch.manual_seed(5)
    dy = _cl(torch.randn(1, n, 1, m, device="cuda", dtype=torch.bfloat16))
    x = _cl(torch.randn(1, k, 1, m, device="cuda", dtype=torch.bfloat16))
    dw = ext.pw_conv2d_bwd_weight_mfma(dy, x)
    ref = torch.einsum("bnhw,bkhw->nk", dy.float(), x.float()).view(n, k, 1, 1)
    tol = 0.5 + 0.02 * (m ** 0.5)
    assert torch.allclose(dw.float(), ref, atol=tol, rtol=0.05), (
        (dw.float() - ref).abs().max().item())


def test_pwconv_stats_epilogue_matches_direct():
    """The forward stats epilogue's folded per-channel sum/sumsq must match a
    direct reduction over the stored y."""
    from deepfake_detection_amd.ops.pwconv import pw_conv2d

    torch.manual_seed(6)
    B, K, N, H = 3, 96, 144, 23
    x = _cl(torch.randn(B, K, H, H, device="cuda", dtype=torch.bfloat16))
    w = torch.randn(N, K, 1, 1, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    y = pw_conv2d(x, w, want_stats=True)
    assert hasattr(y, "_dfd_bn_stats")
    buckets, m, c = y._dfd_bn_stats
    assert m == B * H * H and c == N
    s = buckets[:, 0].sum(dim=0)
    q = buckets[:, 1].sum(dim=0)
    ys = y.detach().float()
    assert torch.allclose(s, ys.sum(dim=(0, 2, 3)), atol=0.5, rtol=1e-3)
    assert torch.allclose(q, (ys * ys).sum(dim=(0, 2, 3)), atol=1.0, rtol=1e-3)


def test_pointwise_module_routes_to_mfma_with_stats():
    """PointwiseConv2d must run the MFMA kernel in bf16 training mode and
    attach producer stats; the fused bn_act consumes them and produces the
    same output/stats as the unfused stats pass."""
    from deepfake_detection_amd.models.layers import PointwiseConv2d
    from deepfake_detection_amd.ops import functional as O

    torch.manual_seed(7)
    conv = PointwiseConv2d(64, 128, 1, bias=False).cuda().to(torch.bfloat16)
    conv.emit_bn_stats = True
    bn = torch.nn.BatchNorm2d(128, momentum=0.01, eps=1e-3).cuda()
    bn2 = torch.nn.BatchNorm2d(128, momentum=0.01, eps=1e-3).cuda()
    x = _cl(torch.randn(4, 64, 17, 17, device="cuda", dtype=torch.bfloat16))

    y = conv(x)
    assert hasattr(y, "_dfd_bn_stats"), "MFMA+stats path not taken"
    out_fused = O.bn_act(y, bn, "silu")
    # unfused stats: same y pushed through bn_act WITHOUT attached stats
    y2 = y.detach().clone()
    out_plain = O.bn_act(y2, bn2, "silu")
    assert torch.allclose(out_fused, out_plain, atol=2e-2, rtol=1e-2)
    assert torch.allclose(bn.running_mean, bn2.running_mean, atol=1e-4, rtol=1e-4)
    assert torch.allclose(bn.running_var, bn2.running_var, atol=1e-4, rtol=1e-4)


def test_ir_block_train_step_with_stats_fusion():
    """One InvertedResidual fwd+bwd in bf16 on the full fused path (MFMA pw,
    producer stats, fused BN) against the CPU fp32 reference block."""
    from deepfake_detection_amd.models.blocks import InvertedResidual

    torch.manual_seed(8)
    blk = InvertedResidual(32, 32, dw_kernel_size=3, exp_ratio=6.0,
                           se_ratio=0.25, act_layer=torch.nn.SiLU).cuda()
    ref = InvertedResidual(32, 32, dw_kernel_size=3, exp_ratio=6.0,
                           se_ratio=0.25, act_layer=torch.nn.SiLU)
    ref.load_state_dict(blk.state_dict())
    blk.train()
    ref.train()
    x = torch.randn(6, 32, 19, 19)
    xg = x.cuda().to(torch.bfloat16).contiguous(
        memory_format=torch.channels_last).requires_grad_(True)
    xr = x.clone().requires_grad_(True)
    with torch.autocast(device_type="cuda", dtype=torch.bfloat16):
        y = blk(xg)
    yr = ref(xr)
    assert torch.allclose(y.detach().float().cpu(), yr.detach(), atol=0.15, rtol=0.1)
    y.float().sum().backward()
    yr.sum().backward()
    assert torch.allclose(xg.grad.float().cpu(), xr.grad, atol=0.3, rtol=0.1)
    g1 = blk.conv_pw.weight.grad.float().cpu()
    g2 = ref.conv_pw.weight.grad
    assert torch.allclose(g1, g2, atol=0.5, rtol=0.1), (g1 - g2).abs().max().item()


def test_bn_act_fused_drop_path():
    """Fused per-sample drop_path scale in bn_act: forward equals
    mask-scaled reference, backward scales dx/dgamma/dbeta and passes the
    residual grad through unscaled."""
    from deepfake_detection_amd.ops.bn_act import fused_bn_act

    torch.manual_seed(9)
    N, C, H, W = 6, 32, 13, 13
    x = _cl(torch.randn(N, C, H, W, device="cuda")).requires_grad_(True)
    res = _cl(torch.randn(N, C, H, W, device="cuda")).requires_grad_(True)
    bn = torch.nn.BatchNorm2d(C, momentum=0.01, eps=1e-3).cuda()
    keep = 0.75
    mask = (torch.rand(N, device="cuda") < keep).float() / keep

    y = fused_bn_act(x, bn.weight, bn.bias, bn.running_mean, bn.running_var,
                     True, 0.01, 1e-3, "silu", res, None, mask)

    rx = x.detach().clone().requires_grad_(True)
    rres = res.detach().clone().requires_grad_(True)
    rm = torch.zeros(C, device="cuda")
    rv = torch.ones(C, device="cuda")
    g = bn.weight.detach().clone().requires_grad_(True)
    b = bn.bias.detach().clone().requires_grad_(True)
    ref = torch.nn.functional.silu(
        torch.nn.functional.batch_norm(rx, rm, rv, g, b, True, 0.01, 1e-3))
    ref = ref * mask.view(-1, 1, 1, 1) + rres
    assert torch.allclose(y, ref, atol=5e-4, rtol=5e-4)

    dy = torch.randn_like(y)
    y.backward(dy)
    ref.backward(dy)
    assert torch.allclose(res.grad, dy)
    assert torch.allclose(x.grad, rx.grad, atol=5e-3, rtol=5e-3)


def test_ir_block_drop_path_stays_fused_and_zeroes_samples():
    """With drop_path_rate>0 in training the IR block must stay on the fused
    GPU path and zero whole samples' block contribution (output == residual
    for dropped samples)."""
    from deepfake_detection_amd.models.blocks import InvertedResidual

    torch.manual_seed(1)
    blk = InvertedResidual(16, 16, exp_ratio=4.0, se_ratio=0.25,
                           act_layer=torch.nn.SiLU, drop_path_rate=0.9).cuda()
    blk.train()
    x = _cl(torch.randn(32, 16, 9, 9, device="cuda", dtype=torch.bfloat16))
    with torch.autocast(device_type="cuda", dtype=torch.bfloat16):
        y = blk(x)
    # with rate 0.9 some samples are certainly dropped: those outputs equal x
    eq = [torch.allclose(y[i].float(), x[i].float(), atol=1e-3)
          for i in range(32)]
    assert any(eq), "no sample dropped at rate 0.9 (mask not applied)"
    assert not all(eq), "all samples dropped (scale wrong)"


def test_dwconv_stats_epilogue_matches_direct(monkeypatch):
    """k3 s1 depthwise stats variant (opt-in): output must match the plain
    kernel and the folded per-channel sums must match a direct reduction
    over y."""
    monkeypatch.setenv("DFD_AMD_DW_STATS", "1")
    from deepfake_detection_amd.ops.dwconv import dw_conv2d

    torch.manual_seed(13)
    B, C, H = 3, 96, 37
    x = _cl(torch.randn(B, C, H, H, device="cuda", dtype=torch.bfloat16))
    w = torch.randn(C, 1, 3, 3, device="cuda", dtype=torch.bfloat16)
    y = dw_conv2d(x, w, None, 1, 1, 1, want_stats=True)
    y0 = dw_conv2d(x, w, None, 1, 1, 1)
    assert torch.equal(y, y0), "stats variant changed the conv output"
    assert hasattr(y, "_dfd_bn_stats")
    buckets, m, c = y._dfd_bn_stats
    assert m == B * H * H and c == C
    ys = y.detach().float()
    assert torch.allclose(buckets[:, 0].sum(0), ys.sum(dim=(0, 2, 3)),
                          atol=0.5, rtol=1e-3)
    assert torch.allclose(buckets[:, 1].sum(0), (ys * ys).sum(dim=(0, 2, 3)),
                          atol=1.0, rtol=1e-3)
    # k5 variant (LDS-staged weight slice) must also be numerically correct
    w5 = torch.randn(C, 1, 5, 5, device="cuda", dtype=torch.bfloat16)
    y5 = dw_conv2d(x, w5, None, 1, 2, 1, want_stats=True)
    y5p = dw_conv2d(x, w5, None, 1, 2, 1)
    assert torch.equal(y5, y5p)
    b5, m5, c5 = y5._dfd_bn_stats
    assert torch.allclose(b5[:, 0].sum(0), y5.float().sum(dim=(0, 2, 3)),
                          atol=0.5, rtol=1e-3)


@pytest.mark.parametrize("cin,n,img", [(3, 48, 61), (12, 256, 38)])
def test_stem_conv_matches_torch(cin, n, img):
    """Implicit-GEMM stem fwd + bwd-weight vs fp32 torch conv (k3 s2 p1)."""
    from deepfake_detection_amd.ops.stemconv import stem_conv2d

    torch.manual_seed(14)
    B = 3
    x = _cl(torch.randn(B, cin, img, img, device="cuda", dtype=torch.bfloat16))
    w = torch.randn(n, cin, 3, 3, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    y = stem_conv2d(x, w, stride=2, padding=1, want_stats=True)
    ref = torch.nn.functional.conv2d(x.float(), w.detach().float(), None, 2, 1)
    assert y.shape == ref.shape
    assert torch.allclose(y.float(), ref, atol=0.25, rtol=0.05), (
        (y.float() - ref).abs().max().item())
    # stats match a direct reduction
    buckets, m, c = y._dfd_bn_stats
    ys = y.detach().float()
    assert torch.allclose(buckets[:, 0].sum(0), ys.sum(dim=(0, 2, 3)),
                          atol=0.5, rtol=1e-3)
    # bwd-weight
    dy = torch.randn_like(y)
    y.backward(dy)
    rw = w.detach().float().requires_grad_(True)
    ref2 = torch.nn.functional.conv2d(x.float(), rw, None, 2, 1)
    ref2.backward(dy.float())
    assert torch.allclose(w.grad.float(), rw.grad, atol=3.0, rtol=0.05), (
        (w.grad.float() - rw.grad).abs().max().item())


def test_stem_module_routes_and_model_runs():
    """B4's conv_stem is a StemConv2d; a full bf16 train microstep runs the
    MFMA stem (no-grad input) and produces finite grads."""
    import deepfake_detection_amd as dfd
    from deepfake_detection_amd.models.layers import StemConv2d

    torch.manual_seed(15)
    m = dfd.create_model("efficientnet_b0", num_classes=2).cuda()
    m = m.to(memory_format=torch.channels_last)
    assert isinstance(m.conv_stem, StemConv2d)
    x = torch.randn(4, 3, 65, 65, device="cuda").to(
        memory_format=torch.channels_last)
    with torch.autocast("cuda", torch.bfloat16):
        out = m(x)
        loss = out.float().sum()
    loss.backward()
    assert torch.isfinite(m.conv_stem.weight.grad).all()


@pytest.mark.parametrize("smoothing", [0.0, 0.1])
def test_fused_head_ce_matches_torch(smoothing):
    """Fused classifier GEMM + smoothed-CE (fwd loss/logits + dx/dW/db) vs
    the eager torch chain."""
    from deepfake_detection_amd.ops.head import fused_head_ce

    torch.manual_seed(16)
    B, F, C = 37, 448, 2
    x = torch.randn(B, F, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    w = torch.randn(C, F, device="cuda", requires_grad=True)
    b = torch.randn(C, device="cuda", requires_grad=True)
    t = torch.randint(0, C, (B,), device="cuda")

    loss, logits = fused_head_ce(x, w, b, t, smoothing)
    rx = x.detach().float().requires_grad_(True)
    rw = w.detach().clone().requires_grad_(True)
    rb = b.detach().clone().requires_grad_(True)
    rlogits = torch.nn.functional.linear(rx, rw, rb)
    logp = torch.nn.functional.log_softmax(rlogits, dim=-1)
    nll = torch.nn.functional.nll_loss(logp, t)
    rloss = (1 - smoothing) * nll + smoothing * (-logp.mean(dim=-1)).mean()
    assert torch.allclose(logits, rlogits, atol=0.2, rtol=0.02)
    assert abs(loss.item() - rloss.item()) < 0.02 * max(1.0, rloss.item())

    loss.backward()
    rloss.backward()
    assert torch.allclose(w.grad, rw.grad, atol=0.02, rtol=0.05)
    assert torch.allclose(b.grad, rb.grad, atol=0.01, rtol=0.05)
    assert torch.allclose(x.grad.float(), rx.grad, atol=0.05, rtol=0.05)


def test_engine_train_epoch_fused_head():
    """One engine train_epoch batch on GPU takes the fused-head path and
    produces finite metrics."""
    import types

    import deepfake_detection_amd as dfd
    from deepfake_detection_amd.engine import _use_fused_head, train_epoch
    from deepfake_detection_amd.optim import RMSpropTF

    torch.manual_seed(17)
    model = dfd.create_model("efficientnet_b0", num_classes=2).cuda()
    model = model.to(memory_format=torch.channels_last)
    opt = RMSpropTF(model.parameters(), lr=1e-5, alpha=0.9, eps=1e-3, momentum=0.9)
    loss_fn = torch.nn.CrossEntropyLoss().cuda()
    x = torch.randn(6, 3, 65, 65, device="cuda")
    t = torch.randint(0, 2, (6,), device="cuda")
    assert _use_fused_head(model, loss_fn, t, use_cuda=True)
    args = types.SimpleNamespace(log_interval=1, prefetcher=True, amp=True,
                                 recovery_interval=0, save_images=False)
    metrics = train_epoch(0, model, [(x, t)], opt, loss_fn, args,
                          torch.device("cuda", 0))
    assert torch.isfinite(torch.tensor(metrics["loss"]))
    assert 0.0 <= metrics["prec1"] <= 100.0


def test_half_model_inference_path():
    """The reference's model_half.pth.tar path (.half() model, test.py:44-47):
    fused BN must accept fp16 BN params in eval (fp32 read-only casts)."""
    import deepfake_detection_amd as dfd

    torch.manual_seed(18)
    m = dfd.create_model("efficientnet_b0", num_classes=2).cuda()
    ref = dfd.create_model("efficientnet_b0", num_classes=2)
    ref.load_state_dict(m.state_dict())
    m = m.half().eval().to(memory_format=torch.channels_last)
    ref = ref.eval()
    x = torch.randn(2, 3, 65, 65)
    xg = x.cuda().half().contiguous(memory_format=torch.channels_last)
    with torch.no_grad():
        y = m(xg)
        yr = ref(x)
    assert torch.isfinite(y.float()).all()
    assert torch.allclose(y.float().cpu(), yr, atol=0.1, rtol=0.05), (
        (y.float().cpu() - yr).abs().max().item())


def test_runner_main_synthetic_gpu(tmp_path, monkeypatch):
    """Full runner main() on GPU: bf16 autocast, fused kernels, prefetcher
    path, EMA, checkpoint written — the most reference-like integration."""
    from deepfake_detection_amd.runners.train import _parse_args, main

    monkeypatch.chdir(tmp_path)
    args, args_text = _parse_args([
        "--synthetic-data", "--synthetic-len", "16", "--model", "efficientnet_lite0",
        "--num-classes", "2", "--input-size-v2", "12,64,64", "-b", "8",
        "--epochs", "1", "--sched", "step", "--decay-epochs", "2",
        "--warmup-epochs", "0", "--opt", "rmsproptf", "--opt-eps", "0.001",
        "--workers", "0", "--log-interval", "1", "--model-ema",
        "--drop-path", "0.2", "--eval-metric", "loss", "--model-version", "tg",
    ])
    main(0, args, args_text)
    out_dir = tmp_path / "output" / "tg-efficientnet_lite0"
    assert (out_dir / "checkpoint-0.pth.tar").exists()
    ck = torch.load(str(out_dir / "checkpoint-0.pth.tar"), weights_only=False)
    assert "state_dict_ema" in ck
    for v in ck["state_dict"].values():
        assert torch.isfinite(v.float()).all()


def test_prefetch_loader_v3_gpu_stream_path():
    """PrefetchLoader_v3 on GPU: side-HIP-stream H2D + fused uint8->bf16
    normalize + GPU RandomErasing produce a correctly normalized
    channels_last batch (the production input path, reference
    loader.py:213-289)."""
    import torch.utils.data as tud

    from deepfake_detection_amd.data.loader import PrefetchLoader_v3, fast_collate

    torch.manual_seed(19)

    class _DS(tud.Dataset):
        def __len__(self):
            return 8

        def __getitem__(self, i):
            import numpy as np

            rng = np.random.RandomState(i)
            return rng.randint(0, 256, (12, 33, 35), dtype=np.uint8), i % 2

    dl = tud.DataLoader(_DS(), batch_size=4, collate_fn=fast_collate)
    pf = PrefetchLoader_v3(dl, img_num=4, dtype=torch.bfloat16, re_prob=0.5)
    batches = list(pf)
    assert len(batches) == 2
    x, t = batches[0]
    assert x.is_cuda and x.dtype == torch.bfloat16
    assert x.is_contiguous(memory_format=torch.channels_last)
    assert x.shape == (4, 12, 33, 35)
    # normalization: uint8 [0,255] -> roughly [-3, 3] after mean/std
    assert x.float().abs().max().item() < 4.0
    assert t.is_cuda
