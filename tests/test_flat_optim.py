"""FlatParams storage and FusedSGD vs torch.optim.SGD."""
import torch

from pytorch_distributed_train_amd.parallel.flat import FlatParams
from pytorch_distributed_train_amd.parallel.optim import FusedSGD


def _tiny_model():
    torch.manual_seed(1)
    return torch.nn.Sequential(
        torch.nn.Linear(8, 16), torch.nn.ReLU(), torch.nn.Linear(16, 4))


def test_flat_views_alias_master():
    m = _tiny_model()
    flat = FlatParams(m.named_parameters())
    for p in m.parameters():
        assert p.data_ptr() >= flat.master.data_ptr()
        assert p.grad is not None
    # mutating master mutates params
    flat.master.add_(1.0)
    for p in m.parameters():
        assert (p.data != 0).all() or p.numel() == 0


def test_flat_reverse_registration_order():
    m = _tiny_model()
    flat = FlatParams(m.named_parameters())
    # last-registered param (2.bias) sits first in the flat buffer
    assert flat.names[0].startswith("2.")
    assert flat.names[-1].startswith("0.")


def test_fused_sgd_matches_torch_sgd():
    torch.manual_seed(2)
    for wd, mom, nesterov in [(0.0, 0.0, False), (1e-4, 0.9, False),
                              (1e-2, 0.9, True)]:
        m1 = _tiny_model()
        m2 = _tiny_model()
        m2.load_state_dict(m1.state_dict())
        flat = FlatParams(m1.named_parameters())
        opt1 = FusedSGD(flat, lr=0.1, momentum=mom, weight_decay=wd,
                        nesterov=nesterov)
        opt2 = torch.optim.SGD(m2.parameters(), lr=0.1, momentum=mom,
                               weight_decay=wd, nesterov=nesterov)
        x = torch.randn(4, 8)
        for _ in range(3):
            opt1.zero_grad()
            opt2.zero_grad()
            m1(x).pow(2).mean().backward()
            m2(x).pow(2).mean().backward()
            opt1.step()
            opt2.step()
        for p1, p2 in zip(m1.parameters(), m2.parameters()):
            assert torch.allclose(p1, p2, atol=1e-6), (p1 - p2).abs().max()


def test_optimizer_state_dict_roundtrip():
    m = _tiny_model()
    flat = FlatParams(m.named_parameters())
    opt = FusedSGD(flat, lr=0.1, momentum=0.9)
    x = torch.randn(4, 8)
    m(x).pow(2).mean().backward()
    opt.step()
    sd = opt.state_dict()
    assert sd["param_groups"][0]["momentum"] == 0.9
    m2 = _tiny_model()
    flat2 = FlatParams(m2.named_parameters())
    opt2 = FusedSGD(flat2, lr=0.1, momentum=0.9)
    opt2.load_state_dict(sd)
    assert torch.allclose(opt2.momentum_buf, opt.momentum_buf)
