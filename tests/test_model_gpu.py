"""End-to-end GPU model tests: our NHWC bf16 model vs the fp32 CPU torch
twin, one full train step, loss decrease, and native-extension presence."""
import pytest
import torch

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def test_extension_is_native():
    from pytorch_distributed_train_amd.ops.backend import ext, has_ext
    assert has_ext()
    assert "ops/_hip_ops" in ext().__file__.replace("\\", "/")


def test_resnet18_forward_close_to_fp32_reference():
    from pytorch_distributed_train_amd.models import build_model
    from tests.torch_twin import twin_resnet18
    torch.manual_seed(0)
    ours = build_model("resnet18", 10, torch.bfloat16).to(DEV)
    twin = twin_resnet18(10)
    twin.load_state_dict({k: v.cpu() for k, v in ours.state_dict().items()})
    ours.eval()
    twin.eval()
    x = torch.randn(4, 3, 64, 64)
    with torch.no_grad():
        y_ours = ours(x.to(DEV)).float().cpu()
        y_twin = twin(x)
    err = (y_ours - y_twin).abs().max().item()
    scale = y_twin.abs().max().item()
    assert err < 0.05 * max(scale, 1.0), (err, scale)


def test_resnet50_train_step_loss_decreases():
    from pytorch_distributed_train_amd.models import build_model
    from pytorch_distributed_train_amd.ops import functional as OF
    from pytorch_distributed_train_amd.parallel.flat import FlatParams
    from pytorch_distributed_train_amd.parallel.optim import FusedSGD
    torch.manual_seed(1)
    model = build_model("resnet50", 1000, torch.bfloat16).to(DEV)
    model.train()
    flat = FlatParams(model.named_parameters(), device=torch.device(DEV))
    opt = FusedSGD(flat, lr=0.02, momentum=0.9)
    x = torch.randn(8, 3, 96, 96, device=DEV)
    y = torch.randint(0, 1000, (8,), device=DEV)
    losses = []
    for _ in range(6):
        opt.zero_grad()
        loss = OF.softmax_cross_entropy(model(x), y)
        loss.backward()
        opt.step()
        losses.append(float(loss))
    assert all(l == l for l in losses), losses  # no NaN
    assert losses[-1] < losses[0], losses


def test_resnet18_grads_close_to_twin():
    from pytorch_distributed_train_amd.models import build_model
    from pytorch_distributed_train_amd.ops import functional as OF
    from tests.torch_twin import twin_resnet18
    torch.manual_seed(2)
    ours = build_model("resnet18", 10, torch.bfloat16).to(DEV)
    twin = twin_resnet18(10)
    twin.load_state_dict({k: v.cpu() for k, v in ours.state_dict().items()})
    ours.train()
    twin.train()
    x = torch.randn(8, 3, 64, 64)
    labels = torch.randint(0, 10, (8,))
    loss_o = OF.softmax_cross_entropy(ours(x.to(DEV)), labels.to(DEV))
    loss_o.backward()
    loss_t = torch.nn.functional.cross_entropy(twin(x), labels)
    loss_t.backward()
    assert abs(float(loss_o) - float(loss_t)) < 0.05 * max(1.0,
                                                           abs(float(loss_t)))
    ours_g = {n: p.grad.float().cpu() for n, p in ours.named_parameters()}
    bad = []
    for n, p in twin.named_parameters():
        ref = p.grad
        err = (ours_g[n] - ref).abs().max().item()
        tol = 0.08 * max(ref.abs().max().item(), 0.1)
        if err > tol:
            bad.append((n, err, tol))
    assert not bad, bad
