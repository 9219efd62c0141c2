"""GPU transformer-kernel numerics vs fp32 torch + ViT end-to-end."""
import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from pytorch_distributed_train_amd.ops.backend import ext

DEV = "cuda:0"


def _rand(*shape, dtype=torch.bfloat16, seed=0):
    g = torch.Generator(device="cpu").manual_seed(seed)
    return torch.randn(*shape, generator=g).to(DEV, dtype)


def _close(a, b, rtol=2e-2, atol=2e-2, what=""):
    a, b = a.float().cpu(), b.float().cpu()
    err = (a - b).abs().max().item()
    tol = atol + rtol * b.abs().max().item()
    assert err <= tol, f"{what}: max err {err} > tol {tol}"


def test_gemm_nt_batched():
    a = _rand(5, 128, 64, seed=1)
    b = _rand(5, 192, 64, seed=2)
    y = ext().gemm_nt_batched(a, b)
    ref = a.float() @ b.float().transpose(1, 2)
    _close(y, ref, what="batched gemm")
    y32 = ext().gemm_nt_batched(a, b, True)
    assert y32.dtype == torch.float32
    _close(y32, ref, rtol=1e-2, what="batched gemm f32")


def test_ln_fwd_bwd():
    x = _rand(37, 768, seed=3)
    w = torch.randn(768, device=DEV)
    b = torch.randn(768, device=DEV)
    y, mean, invstd = ext().ln_fwd(x, w, b, 1e-6)
    x32 = x.float()
    ref = F.layer_norm(x32, (768,), w, b, 1e-6)
    _close(y, ref, what="ln fwd")
    dy = _rand(37, 768, seed=4)
    dx, dw, db = ext().ln_bwd(x, dy, w, mean, invstd)
    xr = x32.detach().requires_grad_()
    wr = w.detach().requires_grad_()
    br = b.detach().requires_grad_()
    F.layer_norm(xr, (768,), wr, br, 1e-6).backward(dy.float())
    _close(dx, xr.grad, what="ln dx")
    _close(dw, wr.grad, rtol=2e-2, atol=5e-2, what="ln dw")
    _close(db, br.grad, rtol=2e-2, atol=5e-2, what="ln db")


def test_gelu_fwd_bwd():
    x = _rand(64, 128, seed=5)
    y = ext().gelu_fwd(x)
    _close(y, F.gelu(x.float()), what="gelu fwd")
    dy = _rand(64, 128, seed=6)
    xr = x.float().detach().requires_grad_()
    F.gelu(xr).backward(dy.float())
    _close(ext().gelu_bwd(x, dy), xr.grad, what="gelu bwd")


def test_masked_softmax():
    rows, n, nv = 64, 256, 197
    s = torch.randn(rows, n, device=DEV) * 3
    p = ext().attn_softmax_fwd(s, nv, 0.125, torch.bfloat16)
    ref = torch.softmax(s[:, :nv].float() * 0.125, dim=-1)
    _close(p[:, :nv], ref, rtol=1e-2, atol=1e-3, what="softmax fwd")
    assert p[:, nv:].abs().max().item() == 0
    dp = _rand(rows, n, seed=7)
    ds = ext().attn_softmax_bwd(p, dp, 0.125)
    p32 = p.float()
    dot = (p32 * dp.float()).sum(-1, keepdim=True)
    ref_ds = 0.125 * p32 * (dp.float() - dot)
    _close(ds, ref_ds, rtol=1e-2, atol=1e-3, what="softmax bwd")


def test_attention_end_to_end_vs_torch():
    from pytorch_distributed_train_amd.ops import functional as OF
    S, NV, HD = 128, 100, 64
    q = _rand(4, S, HD, seed=8).requires_grad_(True)
    k = _rand(4, S, HD, seed=9).requires_grad_(True)
    v = _rand(4, S, HD, seed=10)
    with torch.no_grad():
        v[:, NV:] = 0
    v.requires_grad_(True)
    out = OF.attention(q, k, v, NV, 0.125)
    q2 = q.detach()[:, :NV].float().requires_grad_(True)
    k2 = k.detach()[:, :NV].float().requires_grad_(True)
    v2 = v.detach()[:, :NV].float().requires_grad_(True)
    ref = torch.softmax((q2 @ k2.transpose(1, 2)) * 0.125, -1) @ v2
    _close(out[:, :NV], ref, what="attention fwd")
    g = _rand(4, NV, HD, seed=11).float()
    gpad = torch.zeros(4, S, HD, device=DEV, dtype=torch.bfloat16)
    gpad[:, :NV] = g.to(torch.bfloat16)
    dq, dk, dv = torch.autograd.grad(out, (q, k, v), gpad)
    rq, rk, rv = torch.autograd.grad(ref, (q2, k2, v2), gpad[:, :NV].float())
    _close(dq[:, :NV], rq, rtol=3e-2, atol=3e-2, what="attention dq")
    _close(dk[:, :NV], rk, rtol=3e-2, atol=3e-2, what="attention dk")
    _close(dv[:, :NV], rv, rtol=3e-2, atol=3e-2, what="attention dv")


def test_vit_train_step_gpu():
    from pytorch_distributed_train_amd.models import build_model
    from pytorch_distributed_train_amd.ops import functional as OF
    from pytorch_distributed_train_amd.parallel.flat import FlatParams
    from pytorch_distributed_train_amd.parallel.optim import FusedSGD
    torch.manual_seed(3)
    m = build_model("vit_b_16", 1000, torch.bfloat16).to(DEV)
    m.train()
    flat = FlatParams(m.named_parameters(), device=torch.device(DEV))
    opt = FusedSGD(flat, lr=0.01, momentum=0.9)
    x = torch.randn(8, 3, 224, 224, device=DEV)
    y = torch.randint(0, 1000, (8,), device=DEV)
    losses = []
    for _ in range(8):
        opt.zero_grad()
        loss = OF.softmax_cross_entropy(m(x), y)
        loss.backward()
        opt.step()
        losses.append(float(loss.detach()))
    assert all(l == l for l in losses), losses
    assert min(losses[1:]) < 0.7 * losses[0], losses


def test_attention_qkv_zero_copy_matches_permute_path():
    """attn_fwd/attn_bwd (in-place strided GEMMs) vs the permute-based
    AttentionFn on the same qkv tensor — fwd and d(qkv)."""
    from pytorch_distributed_train_amd.ops import functional as OF
    b, s, h, hd, nv = 2, 128, 4, 64, 100
    scale = 0.125
    qkv = _rand(b * s, 3 * h * hd, seed=20)
    # zero the padded V rows (as the model guarantees via input padding)
    qkv3 = qkv.view(b, s, 3, h, hd)
    with torch.no_grad():
        qkv3[:, nv:, 2] = 0
    qkv_a = qkv.clone().requires_grad_(True)
    qkv_b = qkv.clone().requires_grad_(True)

    out_a = OF.attention_qkv(qkv_a, b, s, h, hd, nv, scale)  # [b,s,h*hd]

    qp = qkv_b.view(b, s, 3, h, hd).permute(2, 0, 3, 1, 4)
    q = qp[0].reshape(b * h, s, hd).contiguous()
    k = qp[1].reshape(b * h, s, hd).contiguous()
    v = qp[2].reshape(b * h, s, hd).contiguous()
    out_ref = OF.attention(q, k, v, nv, scale)
    out_ref = out_ref.view(b, h, s, hd).permute(0, 2, 1, 3).reshape(
        b, s, h * hd)

    _close(out_a[:, :nv], out_ref[:, :nv], what="attn_qkv fwd")

    g = torch.zeros_like(out_a)
    g[:, :nv] = _rand(b, nv, h * hd, seed=21)
    out_a.backward(g)
    out_ref.backward(g.clone())
    _close(qkv_a.grad, qkv_b.grad, rtol=3e-2, atol=3e-2, what="attn_qkv dqkv")


def test_attention_qkv_shape_sweep_vs_fp32():
    """The fused kernel accepts any 64 <= s <= 256 with head_dim 64 and
    an UNPADDED qkv (attn.hip: clamped row reads, zero prob tails, K=SP
    dK/dV against those tails). Sweep (s, nvalid) shapes and compare fwd
    + d(qkv) against a pure fp32 torch reference with -inf key masking.
    Rows >= nvalid must come back EXACTLY zero-gradient on both sides."""
    from pytorch_distributed_train_amd.ops import functional as OF
    b, h, hd = 2, 4, 64
    d = h * hd
    for i, (s, nv) in enumerate(
            [(64, 64), (100, 97), (197, 197), (256, 197), (256, 256)]):
        scale = 1.0 / hd ** 0.5
        qkv = _rand(b * s, 3 * d, seed=40 + i)
        qkv_a = qkv.clone().requires_grad_(True)
        out = OF.attention_qkv(qkv_a, b, s, h, hd, nv, scale)
        g = torch.zeros(b, s, d, device=DEV, dtype=qkv.dtype)
        g[:, :nv] = _rand(b, nv, d, seed=60 + i)
        out.backward(g.view(b, s, d))

        ref = qkv.float().cpu().view(b, s, 3, h, hd).permute(
            2, 0, 3, 1, 4).clone().requires_grad_(True)  # [3,b,h,s,hd]
        q, k, v = ref[0], ref[1], ref[2]
        sc = (q @ k.transpose(-1, -2)) * scale
        sc[..., nv:] = float("-inf")
        p = sc.softmax(-1)
        o = (p @ v).permute(0, 2, 1, 3).reshape(b, s, d)
        o.backward(g.float().cpu())

        _close(out.view(b, s, d)[:, :nv], o[:, :nv],
               what=f"fwd s={s} nv={nv}")
        dref = ref.grad.permute(1, 3, 0, 2, 4).reshape(b * s, 3 * d)
        _close(qkv_a.grad, dref, rtol=3e-2, atol=3e-2,
               what=f"dqkv s={s} nv={nv}")
        if nv < s:
            tail = qkv_a.grad.view(b, s, 3 * d)[:, nv:]
            assert tail.abs().max().item() == 0.0, f"tail s={s} nv={nv}"


def test_pack_transpose_kernel():
    """LDS-tiled [N][K] fp32 -> [K][N] dtype transpose (packs.hip
    pack_transpose_k) vs torch .t(), including non-64-multiple edges."""
    for (n, k) in [(64, 64), (72, 100), (1000, 768), (2304, 768)]:
        master = torch.randn(n * k, device=DEV, dtype=torch.float32)
        pool = torch.full((k * n + 8,), float("nan"), device=DEV,
                          dtype=torch.bfloat16)
        desc = torch.tensor([[0, 0, n, k]], dtype=torch.long, device=DEV)
        tiles = ((n + 63) // 64) * ((k + 63) // 64)
        tprefix = torch.tensor([tiles], dtype=torch.long, device=DEV)
        ext().pack_transpose(master, pool, desc, tprefix, tiles)
        got = pool[:k * n].view(k, n).float()
        ref = master.view(n, k).t().to(torch.bfloat16).float()
        assert torch.equal(got, ref), (n, k)
        assert pool[k * n:].isnan().all(), "wrote past the segment"
