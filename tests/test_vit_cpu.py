"""ViT path on CPU: op-level oracles vs plain torch + end-to-end training."""
import torch
import torch.nn.functional as F

from pytorch_distributed_train_amd.ops import functional as OF


def test_layer_norm_matches_torch():
    torch.manual_seed(0)
    x = torch.randn(4, 7, 64, requires_grad=True)
    w = torch.randn(64, requires_grad=True)
    b = torch.randn(64, requires_grad=True)
    y = OF.layer_norm(x, w, b, 1e-6)
    ref = F.layer_norm(x, (64,), w, b, 1e-6)
    assert torch.allclose(y, ref, atol=1e-5)
    g = torch.randn_like(y)
    dx, dw, db = torch.autograd.grad(y, (x, w, b), g, retain_graph=True)
    rx, rw, rb = torch.autograd.grad(ref, (x, w, b), g)
    assert torch.allclose(dx, rx, atol=1e-5)
    assert torch.allclose(dw, rw, atol=1e-4)
    assert torch.allclose(db, rb, atol=1e-4)


def test_gelu_matches_torch():
    x = torch.randn(300, requires_grad=True)
    y = OF.gelu(x)
    ref = F.gelu(x)
    assert torch.allclose(y, ref, atol=1e-6)
    g = torch.randn_like(y)
    dx = torch.autograd.grad(y, x, g, retain_graph=True)[0]
    rx = torch.autograd.grad(ref, x, g)[0]
    assert torch.allclose(dx, rx, atol=1e-5)


def test_attention_padding_invariance():
    """Padded attention (S=128, nvalid=100) must equal unpadded torch
    attention on the first 100 rows — fwd AND grads."""
    torch.manual_seed(1)
    S, NV, HD = 128, 100, 64
    q = torch.randn(3, S, HD)
    k = torch.randn(3, S, HD)
    v = torch.randn(3, S, HD)
    v[:, NV:] = 0
    q.requires_grad_(True)
    k.requires_grad_(True)
    v.requires_grad_(True)
    out = OF.attention(q, k, v, NV, 0.125)
    # unpadded torch reference
    q2 = q.detach()[:, :NV].clone().requires_grad_(True)
    k2 = k.detach()[:, :NV].clone().requires_grad_(True)
    v2 = v.detach()[:, :NV].clone().requires_grad_(True)
    ref = torch.softmax((q2 @ k2.transpose(1, 2)) * 0.125, -1) @ v2
    assert torch.allclose(out[:, :NV], ref, atol=1e-5), \
        (out[:, :NV] - ref).abs().max()
    g = torch.randn(3, NV, HD)
    gpad = torch.zeros(3, S, HD)
    gpad[:, :NV] = g
    dq, dk, dv = torch.autograd.grad(out, (q, k, v), gpad)
    rq, rk, rv = torch.autograd.grad(ref, (q2, k2, v2), g)
    assert torch.allclose(dq[:, :NV], rq, atol=1e-5)
    assert torch.allclose(dk[:, :NV], rk, atol=1e-5)
    assert torch.allclose(dv[:, :NV], rv, atol=1e-5)
    # grads of padded KEY/VALUE rows must be zero (no leakage)
    assert dk[:, NV:].abs().max() == 0
    assert dv[:, NV:].abs().max() == 0


def test_vit_trains_cpu():
    from pytorch_distributed_train_amd.models import build_model
    from pytorch_distributed_train_amd.parallel.flat import FlatParams
    from pytorch_distributed_train_amd.parallel.optim import FusedSGD
    torch.manual_seed(2)
    # tiny ViT-shaped config via the full class (2 layers to keep it fast)
    from pytorch_distributed_train_amd.models.vit import VisionTransformer
    m = VisionTransformer(num_layers=2, num_classes=10,
                          compute_dtype=torch.float32)
    m.train()
    flat = FlatParams(m.named_parameters())
    opt = FusedSGD(flat, lr=0.05, momentum=0.9)
    x = torch.randn(2, 3, 224, 224)
    y = torch.randint(0, 10, (2,))
    losses = []
    for _ in range(4):
        opt.zero_grad()
        loss = OF.softmax_cross_entropy(m(x), y)
        loss.backward()
        opt.step()
        losses.append(float(loss.detach()))
    assert losses[-1] < losses[0], losses
