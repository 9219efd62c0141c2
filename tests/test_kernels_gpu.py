"""GPU kernel numerics vs the plain PyTorch fp32 reference (the CPU path
of ops.functional). Every test is @pytest.mark.gpu (needs a real MI355X).

Tolerances: bf16 storage => abs err ~ 1e-2 * magnitude on outputs; fp32
outputs (stats, dW) compare at ~1e-3 relative. GEMM checks use asymmetric
operands (transpose-detecting — guide §5.4 rule 16).
"""
import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from pytorch_distributed_train_amd.ops.backend import ext

DEV = "cuda:0"


def _rand(*shape, dtype=torch.bfloat16, seed=0):
    g = torch.Generator(device="cpu").manual_seed(seed)
    return torch.randn(*shape, generator=g).to(DEV, dtype)


def _close(a, b, rtol=2e-2, atol=2e-2, what=""):
    a = a.float().cpu()
    b = b.float().cpu()
    err = (a - b).abs().max().item()
    tol = atol + rtol * b.abs().max().item()
    assert err <= tol, f"{what}: max err {err} > tol {tol}"


# ---------------------------------------------------------------- gemm --
def test_gemm_nt_identity_asymmetric():
    # A = I: C must equal B^T rows exactly (transpose-detecting)
    K = 128
    a = torch.eye(K).to(DEV, torch.bfloat16)
    b = torch.arange(K * K, dtype=torch.float32).reshape(K, K)
    b = ((b % 37) / 37.0 - 0.5).to(DEV, torch.bfloat16)
    c = ext().gemm_nt(a, b)
    _close(c, b.t(), what="gemm I*B^T")


@pytest.mark.parametrize("m,n,k", [(128, 128, 64), (256, 512, 128),
                                   (200, 100, 192), (1000, 64, 256),
                                   (2048, 768, 768), (4096, 256, 192),
                                   (1100, 130, 128)])
def test_gemm_nt_random(m, n, k):
    a = _rand(m, k, seed=1)
    b = _rand(n, k, seed=2)
    ref = a.float() @ b.float().t()
    c = ext().gemm_nt(a, b)
    _close(c, ref, what=f"gemm {m}x{n}x{k}")


def test_gemm_nt_bias_and_f32out():
    a = _rand(64, 64, seed=3)
    b = _rand(192, 64, seed=4)
    bias = torch.randn(192, device=DEV)
    ref = a.float() @ b.float().t() + bias
    c = ext().gemm_nt(a, b, bias)
    _close(c, ref, what="gemm+bias")
    c32 = ext().gemm_nt(a, b, None, True)
    assert c32.dtype == torch.float32
    _close(c32, a.float() @ b.float().t(), rtol=1e-2, what="gemm f32out")


def test_gemm_fp16():
    a = _rand(128, 128, dtype=torch.float16, seed=5)
    b = _rand(128, 128, dtype=torch.float16, seed=6)
    _close(ext().gemm_nt(a, b), a.float() @ b.float().t(), what="gemm fp16")


# ---------------------------------------------------------------- conv --
def _conv_ref(x, w, stride, pad):
    import torch.nn.functional as F
    y = F.conv2d(x.permute(0, 3, 1, 2).float(), w.float(), stride=stride,
                 padding=pad)
    return y.permute(0, 2, 3, 1).contiguous()


@pytest.mark.parametrize("n,h,c,k,r,stride", [
    (2, 14, 64, 64, 1, 1), (2, 14, 64, 128, 1, 2), (2, 14, 64, 64, 3, 1),
    (2, 14, 128, 256, 3, 2), (1, 7, 2048, 512, 1, 1),
    # big-tile path (M >= 4096) with NON-power-of-two spatial dims
    (2, 46, 64, 256, 3, 1), (3, 46, 64, 512, 1, 2), (2, 56, 128, 256, 1, 1)])
def test_conv_fwd(n, h, c, k, r, stride):
    pad = r // 2
    x = _rand(n, h, h, c, seed=7)
    w = (_rand(k, c, r, r, seed=8).float() / (c ** 0.5)).to(torch.bfloat16)
    wt = w.permute(0, 2, 3, 1).contiguous()
    y = ext().conv2d_fwd(x, wt, stride, pad)[0]
    _close(y, _conv_ref(x, w, stride, pad), what=f"conv {c}->{k} r{r}s{stride}")


@pytest.mark.parametrize("n,h,c,k,r,stride", [
    (2, 14, 64, 64, 3, 1), (2, 14, 64, 128, 1, 2), (2, 14, 128, 64, 3, 2),
    # big-tile path with non-power-of-two spatial dims
    (2, 46, 256, 64, 3, 1), (2, 46, 256, 128, 1, 2)])
def test_conv_bwd_data(n, h, c, k, r, stride):
    import torch.nn.functional as F
    pad = r // 2
    p = (h + 2 * pad - r) // stride + 1
    dy = _rand(n, p, p, k, seed=9)
    w = (_rand(k, c, r, r, seed=10).float() / (c ** 0.5)).to(torch.bfloat16)
    w2 = w.permute(2, 3, 1, 0).contiguous()
    dx = ext().conv2d_bwd_data(dy, w2, h, h, stride, pad)
    ref = torch.nn.grad.conv2d_input(
        (n, c, h, h), w.float(), dy.permute(0, 3, 1, 2).float(),
        stride=stride, padding=pad).permute(0, 2, 3, 1)
    _close(dx, ref, what=f"bwd_data {c}<-{k} r{r}s{stride}")


@pytest.mark.parametrize("n,h,c,k,r,stride", [
    (2, 14, 64, 64, 3, 1), (2, 14, 64, 128, 1, 2), (2, 8, 128, 128, 3, 2),
    # big-tile wrw (Kout,C % 256) incl. gather + non-pow2 spatial
    (2, 20, 256, 256, 3, 1), (2, 23, 256, 512, 1, 2), (1, 30, 512, 256, 1, 1)])
def test_conv_bwd_weight(n, h, c, k, r, stride):
    pad = r // 2
    p = (h + 2 * pad - r) // stride + 1
    x = _rand(n, h, h, c, seed=11)
    dy = _rand(n, p, p, k, seed=12)
    dw = ext().conv2d_bwd_weight(x, dy, r, r, stride, pad)
    ref = torch.nn.grad.conv2d_weight(
        x.permute(0, 3, 1, 2).float(), (k, c, r, r),
        dy.permute(0, 3, 1, 2).float(), stride=stride, padding=pad)
    _close(dw, ref, rtol=2e-2, what=f"bwd_weight {c}->{k}")


def test_stem_conv_fwd_and_wrw():
    # C=3 7x7/s2 — the direct stem kernels
    x = _rand(2, 32, 32, 3, seed=13)
    w = (_rand(64, 3, 7, 7, seed=14).float() / 3.0).to(torch.bfloat16)
    wt = w.permute(0, 2, 3, 1).contiguous()
    y = ext().conv2d_fwd(x, wt, 2, 3)[0]
    _close(y, _conv_ref(x, w, 2, 3), what="stem fwd")
    p = y.shape[1]
    dy = _rand(2, p, p, 64, seed=15)
    dw = ext().conv2d_bwd_weight(x, dy, 7, 7, 2, 3)
    ref = torch.nn.grad.conv2d_weight(
        x.permute(0, 3, 1, 2).float(), (64, 3, 7, 7),
        dy.permute(0, 3, 1, 2).float(), stride=2, padding=3)
    _close(dw, ref, what="stem wrw")


# ------------------------------------------------------------------ bn --
def test_bn_stats_apply_bwd():
    n, h, c = 4, 10, 128
    x = _rand(n, h, h, c, seed=16)
    mean, var = ext().bn_stats(x)
    x32 = x.float()
    _close(mean, x32.mean(dim=(0, 1, 2)), rtol=1e-3, atol=1e-3, what="mean")
    _close(var, x32.var(dim=(0, 1, 2), unbiased=False), rtol=1e-3, atol=1e-3,
           what="var")
    w = torch.randn(c, device=DEV)
    b = torch.randn(c, device=DEV)
    invstd = (var + 1e-5).rsqrt()
    y = ext().bn_apply(x, w, b, mean, invstd, True)
    ref = ((x32 - mean) * invstd * w + b).clamp(min=0)
    _close(y, ref, what="bn_apply+relu")

    dy = _rand(n, h, h, c, seed=17)
    sum_dy, sum_dy_xmu = ext().bn_bwd_reduce(x, dy, None, mean)
    dy32 = dy.float()
    _close(sum_dy, dy32.sum(dim=(0, 1, 2)), rtol=1e-2, atol=1e-2,
           what="sum_dy")
    _close(sum_dy_xmu, (dy32 * (x32 - mean)).sum(dim=(0, 1, 2)), rtol=1e-2,
           atol=5e-2, what="sum_dy_xmu")
    cnt = float(n * h * h)
    dx = ext().bn_bwd_elemt(x, dy, None, mean, invstd, w, sum_dy,
                            sum_dy_xmu, cnt)[0]
    ref_dx = (dy32 - sum_dy / cnt -
              (x32 - mean) * invstd * invstd * sum_dy_xmu / cnt) * w * invstd
    _close(dx, ref_dx, what="bn_bwd_elemt")

    # fused ReLU mask path: mask dy by (y > 0) inside both kernels
    s_m, sx_m = ext().bn_bwd_reduce(x, dy, y, mean)
    dym = dy32 * (y.float() > 0)
    _close(s_m, dym.sum(dim=(0, 1, 2)), rtol=1e-2, atol=1e-2,
           what="sum_dy masked")
    _close(sx_m, (dym * (x32 - mean)).sum(dim=(0, 1, 2)), rtol=1e-2,
           atol=5e-2, what="sum_dy_xmu masked")
    dx_m, dres = ext().bn_bwd_elemt(x, dy, y, mean, invstd, w, s_m, sx_m,
                                    cnt, True)
    ref_dx_m = (dym - s_m / cnt -
                (x32 - mean) * invstd * invstd * sx_m / cnt) * w * invstd
    _close(dx_m, ref_dx_m, what="bn_bwd_elemt masked")
    _close(dres, dym, what="bn_bwd_elemt dres")

    # fused residual: y = relu(bn(x) + res)
    res = _rand(n, h, h, c, seed=31)
    yr = ext().bn_apply(x, w, b, mean, invstd, True, res)
    ref_yr = ((x32 - mean) * invstd * w + b + res.float()).clamp(min=0)
    _close(yr, ref_yr, what="bn_apply+res+relu")


def test_bn_stats_train_updates_running():
    n, h, c = 2, 8, 64
    x = _rand(n, h, h, c, seed=30)
    rm = torch.zeros(c, device=DEV)
    rv = torch.ones(c, device=DEV)
    mean, invstd = ext().bn_stats_train(x, rm, rv, 0.1, 1e-5)
    x32 = x.float()
    ref_mean = x32.mean(dim=(0, 1, 2))
    ref_var = x32.var(dim=(0, 1, 2), unbiased=False)
    _close(mean, ref_mean, rtol=1e-3, atol=1e-3, what="train mean")
    _close(invstd, (ref_var + 1e-5).rsqrt(), rtol=1e-3, atol=1e-3,
           what="train invstd")
    nn = n * h * h
    _close(rm, 0.1 * ref_mean, rtol=1e-3, atol=1e-3, what="running_mean")
    _close(rv, 0.9 + 0.1 * ref_var * nn / (nn - 1), rtol=1e-3, atol=1e-3,
           what="running_var")


# ---------------------------------------------------------- elementwise --
def test_relu_add_mask():
    x = _rand(3, 8, 8, 64, seed=18)
    y = ext().relu_fwd(x)
    _close(y, x.float().clamp(min=0), what="relu")
    a, b = _rand(1, 4, 4, 64, seed=19), _rand(1, 4, 4, 64, seed=20)
    _close(ext().add_relu_fwd(a, b), (a.float() + b.float()).clamp(min=0),
           what="add_relu")
    dy = _rand(3, 8, 8, 64, seed=21)
    _close(ext().relu_bwd_mask(dy, y), dy.float() * (y.float() > 0),
           what="relu_bwd")


# --------------------------------------------------------------- pool --
def test_maxpool():
    import torch.nn.functional as F
    x = _rand(2, 12, 12, 64, seed=22)
    y, idx = ext().maxpool_fwd(x, 3, 2, 1)
    ref = F.max_pool2d(x.permute(0, 3, 1, 2).float(), 3, 2, 1)
    _close(y, ref.permute(0, 2, 3, 1), what="maxpool fwd")
    dy = _rand(*y.shape, seed=23)
    dx = ext().maxpool_bwd(dy, idx, 12, 12, 3, 2, 1)
    xr = x.permute(0, 3, 1, 2).float().requires_grad_()
    yr = F.max_pool2d(xr, 3, 2, 1)
    yr.backward(dy.permute(0, 3, 1, 2).float())
    _close(dx, xr.grad.permute(0, 2, 3, 1), what="maxpool bwd")


def test_global_avgpool():
    x = _rand(3, 7, 7, 512, seed=24)
    y = ext().avgpool_global_fwd(x)
    _close(y, x.float().mean(dim=(1, 2)), what="gap fwd")
    dy = _rand(3, 512, seed=25)
    dx = ext().avgpool_global_bwd(dy, 7, 7)
    ref = (dy.float() / 49).view(3, 1, 1, 512).expand(3, 7, 7, 512)
    _close(dx, ref, what="gap bwd")


# ----------------------------------------------------------------- ce --
def test_softmax_ce():
    import torch.nn.functional as F
    b, n = 37, 1000
    logits = _rand(b, n, seed=26)
    labels = torch.randint(0, n, (b,), device=DEV)
    loss_per, lse = ext().softmax_ce_fwd(logits, labels, 0.0)
    l32 = logits.float()
    ref_lse = torch.logsumexp(l32, dim=1)
    _close(lse, ref_lse, rtol=1e-3, atol=1e-3, what="lse")
    ref_loss = F.cross_entropy(l32, labels, reduction="none")
    _close(loss_per, ref_loss, rtol=1e-3, atol=1e-3, what="ce loss")
    gout = torch.ones(1, device=DEV)
    d = ext().softmax_ce_bwd(logits, labels, lse, 0.0, 1.0 / b, gout)
    lr = l32.detach().requires_grad_()
    F.cross_entropy(lr, labels).backward()
    _close(d, lr.grad, rtol=1e-2, atol=1e-3, what="ce bwd")


def test_softmax_ce_label_smoothing():
    import torch.nn.functional as F
    b, n = 16, 1000
    logits = _rand(b, n, seed=27)
    labels = torch.randint(0, n, (b,), device=DEV)
    loss_per, lse = ext().softmax_ce_fwd(logits, labels, 0.1)
    ref = F.cross_entropy(logits.float(), labels, reduction="none",
                          label_smoothing=0.1)
    _close(loss_per, ref, rtol=1e-3, atol=1e-3, what="ce smooth")


# ---------------------------------------------------------------- sgd --
def test_fused_sgd_matches_torch():
    n = 4096
    g0 = torch.randn(n, device=DEV)
    p1 = torch.randn(n, device=DEV)
    p2 = p1.clone()
    m1 = torch.zeros(n, device=DEV)
    ref_p = p2.clone().requires_grad_()
    opt = torch.optim.SGD([ref_p], lr=0.1, momentum=0.9, weight_decay=1e-4)
    for i in range(3):
        g = g0 * (i + 1)
        ext().fused_sgd(p1, g, m1, 0.1, 0.9, 1e-4, 0.0, False, 1.0)
        ref_p.grad = g.clone()
        opt.step()
    _close(p1, ref_p.detach(), rtol=1e-5, atol=1e-5, what="fused sgd")


def test_unscale_check():
    g = torch.randn(512, device=DEV) * 4.0
    found = ext().unscale_check(g, 0.25)
    assert float(found) == 0.0
    g2 = torch.randn(512, device=DEV)
    g2[100] = float("inf")
    found2 = ext().unscale_check(g2, 1.0)
    assert float(found2) == 1.0


@pytest.mark.gpu
@pytest.mark.parametrize("shape", [
    # (N, HW, C, K, ksize, stride) — small-tile, big-tile, and stem-ish
    (2, 30, 64, 128, 3, 1),
    (2, 46, 64, 256, 1, 1),
    (3, 23, 128, 512, 3, 2),
])
def test_conv_fwd_fused_bn_stats(shape):
    """conv2d_fwd(want_bn_stats=True) epilogue partials finalize to the
    same mean/invstd as the standalone bn_stats_train pass over y."""
    N, HW, C, K, ks, stride = shape
    torch.manual_seed(5)
    dev = "cuda"
    x = torch.randn(N, HW, HW, C, device=dev, dtype=torch.bfloat16)
    w = torch.randn(K, C, ks, ks, device=dev) * 0.05
    wt = w.permute(0, 2, 3, 1).to(torch.bfloat16).contiguous()
    pad = ks // 2
    y, part = ext().conv2d_fwd(x, wt, stride, pad, True)
    rows = y.numel() // K
    rm1 = torch.zeros(K, device=dev)
    rv1 = torch.ones(K, device=dev)
    mean1, invstd1 = ext().bn_stats_train_from_part(
        part, rows, rm1, rv1, 0.1, 1e-5)
    rm2 = torch.zeros(K, device=dev)
    rv2 = torch.ones(K, device=dev)
    mean2, invstd2 = ext().bn_stats_train(y, rm2, rv2, 0.1, 1e-5)
    # the epilogue sums f32 accumulators pre-cast; bn_stats_train reads the
    # bf16-stored y — equal to bf16 rounding
    assert torch.allclose(mean1, mean2, atol=5e-3, rtol=1e-2), \
        (mean1 - mean2).abs().max()
    assert torch.allclose(invstd1, invstd2, atol=5e-3, rtol=2e-2), \
        (invstd1 - invstd2).abs().max()
    assert torch.allclose(rm1, rm2, atol=5e-3, rtol=1e-2)
    assert torch.allclose(rv1, rv2, atol=5e-3, rtol=2e-2)


@pytest.mark.gpu
def test_bn_bwd_premask_matches_stored_relu_mask():
    """MASK=2 (gamma*xhat+beta > 0 recomputed in-kernel) must equal MASK=1
    (reading the stored relu output) away from the bf16 rounding boundary."""
    torch.manual_seed(7)
    dev = "cuda"
    N, HW, C = 3, 17, 128
    x = torch.randn(N, HW, HW, C, device=dev, dtype=torch.bfloat16)
    dy = torch.randn_like(x)
    mean, var = ext().bn_stats(x)
    invstd = (var + 1e-5).rsqrt()
    w = torch.randn(C, device=dev) * 0.5 + 1.0
    b = torch.randn(C, device=dev) * 0.1
    y = ext().bn_apply(x, w, b, mean, invstd, True, None)
    s1, sx1 = ext().bn_bwd_reduce(x, dy, y, mean)
    s2, sx2 = ext().bn_bwd_reduce(x, dy, None, mean, invstd=invstd,
                                  weight=w, bias=b)
    assert torch.allclose(s1, s2, atol=1e-1, rtol=1e-3), \
        (s1 - s2).abs().max()
    assert torch.allclose(sx1, sx2, atol=1e-1, rtol=1e-3)
    n = float(N * HW * HW)
    dx1 = ext().bn_bwd_elemt(x, dy, y, mean, invstd, w, s1, sx1, n,
                             False)[0]
    dx2 = ext().bn_bwd_elemt(x, dy, None, mean, invstd, w, s1, sx1, n,
                             False, bias=b)[0]
    diff = (dx1.float() - dx2.float()).abs()
    assert (diff > 0).float().mean() < 1e-3  # only boundary elements differ
