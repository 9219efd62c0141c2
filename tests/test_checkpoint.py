"""Checkpoint save/resume roundtrip."""
import torch

from pytorch_distributed_train_amd.models import build_model
from pytorch_distributed_train_amd.parallel.flat import FlatParams
from pytorch_distributed_train_amd.parallel.optim import FusedSGD
from pytorch_distributed_train_amd.ops import functional as OF
from pytorch_distributed_train_amd.utils.checkpoint import (load_checkpoint,
                                                            save_checkpoint)


def _train_steps(model, opt, n):
    torch.manual_seed(3)
    x = torch.randn(4, 3, 32, 32)
    y = torch.randint(0, 10, (4,))
    for _ in range(n):
        opt.zero_grad()
        OF.softmax_cross_entropy(model(x), y).backward()
        opt.step()


def test_save_resume_roundtrip(tmp_path):
    torch.manual_seed(4)
    model = build_model("resnet18", 10, torch.float32)
    model.train()
    flat = FlatParams(model.named_parameters())
    opt = FusedSGD(flat, lr=0.01, momentum=0.9)
    _train_steps(model, opt, 2)
    path = str(tmp_path / "ckpt.pt")
    save_checkpoint(path, epoch=1, step=2, model=model, optimizer=opt)

    model2 = build_model("resnet18", 10, torch.float32)
    model2.train()
    flat2 = FlatParams(model2.named_parameters())
    opt2 = FusedSGD(flat2, lr=0.01, momentum=0.9)
    state = load_checkpoint(path, model2, opt2)
    assert state["epoch"] == 1 and state["step"] == 2
    assert torch.allclose(flat.master, flat2.master)
    assert torch.allclose(opt.momentum_buf, opt2.momentum_buf)
    # continued training must be bit-identical
    _train_steps(model, opt, 2)
    _train_steps(model2, opt2, 2)
    assert torch.equal(flat.master, flat2.master)


def test_module_prefix_stripped(tmp_path):
    model = build_model("resnet18", 10, torch.float32)
    sd = {"module." + k: v for k, v in model.state_dict().items()}
    path = str(tmp_path / "pref.pt")
    torch.save({"epoch": 0, "model": sd}, path)
    model2 = build_model("resnet18", 10, torch.float32)
    load_checkpoint(path, model2)
    for a, b in zip(model.state_dict().values(), model2.state_dict().values()):
        assert torch.equal(a, b)
