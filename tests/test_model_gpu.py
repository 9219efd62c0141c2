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
    opt = FusedSGD(flat, lr=0.005, momentum=0.9)
    x = torch.randn(8, 3, 96, 96, device=DEV)
    y = torch.randint(0, 1000, (8,), device=DEV)
    losses = []
    for _ in range(10):
        opt.zero_grad()
        loss = OF.softmax_cross_entropy(model(x), y)
        loss.backward()
        opt.step()
        losses.append(float(loss.detach()))
    assert all(l == l for l in losses), losses  # no NaN
    assert min(losses[1:]) < 0.6 * losses[0], losses


def test_resnet18_grads_competitive_with_torch_bf16():
    """bf16 error amplifies through the deep backward chain, so comparing
    straight against fp32 fails for ANY bf16 implementation (torch
    autocast shows the same per-layer error — see
    profiles/diag_grads_r01.log). The correct oracle: our per-layer grad
    error vs the fp32 twin must be within 3x of torch-autocast-bf16's own
    error on the same data (plus a small floor)."""
    import torch.nn.functional as F
    from pytorch_distributed_train_amd.models import build_model
    from pytorch_distributed_train_amd.ops import functional as OF
    from tests.torch_twin import twin_resnet18
    torch.manual_seed(2)
    ours = build_model("resnet18", 10, torch.bfloat16).to(DEV)
    sd = {k: v.clone() for k, v in ours.state_dict().items()}
    twin32 = twin_resnet18(10).to(DEV)
    twin32.load_state_dict(sd)
    twin_ac = twin_resnet18(10).to(DEV)
    twin_ac.load_state_dict(sd)
    ours.train()
    twin32.train()
    twin_ac.train()
    x = torch.randn(8, 3, 64, 64, device=DEV)
    labels = torch.randint(0, 10, (8,), device=DEV)

    loss_o = OF.softmax_cross_entropy(ours(x), labels)
    loss_o.backward()
    loss_t = F.cross_entropy(twin32(x), labels)
    loss_t.backward()
    with torch.autocast("cuda", torch.bfloat16):
        loss_a = F.cross_entropy(twin_ac(x), labels)
    loss_a.backward()
    assert abs(float(loss_o) - float(loss_t)) < 0.05 * max(
        1.0, abs(float(loss_t)))

    g_o = {n: p.grad.float() for n, p in ours.named_parameters()}
    g_ac = {n: p.grad.float() for n, p in twin_ac.named_parameters()}
    bad = []
    for n, p in twin32.named_parameters():
        ref = p.grad.float()
        err_ours = (g_o[n] - ref).abs().max().item()
        err_ac = (g_ac[n] - ref).abs().max().item()
        tol = 3.0 * err_ac + 0.02 * max(ref.abs().max().item(), 0.05)
        if err_ours > tol:
            bad.append((n, err_ours, err_ac))
    assert not bad, bad


def test_rccl_single_rank_roundtrip():
    """RCCL binding: comm init + allreduce/broadcast/allgather with
    world_size 1 (the 8-GPU path is driver-measured; this validates the
    binding, stream glue and dtype mapping on real hardware)."""
    from pytorch_distributed_train_amd.ops.backend import ext
    uid = ext().rccl_unique_id()
    comm = ext().rccl_comm_init(uid, 0, 1)
    s = torch.cuda.current_stream().cuda_stream
    t = torch.arange(1024, dtype=torch.float32, device=DEV)
    ref = t.clone()
    ext().rccl_allreduce_avg(t, comm, s)
    torch.cuda.synchronize()
    assert torch.equal(t, ref)
    ext().rccl_broadcast(t, 0, comm, s)
    g = torch.empty(1024, dtype=torch.float32, device=DEV)
    ext().rccl_allgather(t, g, comm, s)
    torch.cuda.synchronize()
    assert torch.equal(g, ref)
    ext().rccl_comm_destroy(comm)


def test_fp16_gradscaler_path():
    """ResNet-18 fp16 + GradScaler (BASELINE config #4 machinery): trains
    without NaNs; an injected inf gradient skips the step and backs off
    the scale."""
    from pytorch_distributed_train_amd.models import build_model
    from pytorch_distributed_train_amd.ops import functional as OF
    from pytorch_distributed_train_amd.parallel.flat import FlatParams
    from pytorch_distributed_train_amd.parallel.optim import FusedSGD
    from pytorch_distributed_train_amd.utils.amp import GradScaler
    torch.manual_seed(5)
    m = build_model("resnet18", 10, torch.float16).to(DEV)
    m.train()
    flat = FlatParams(m.named_parameters(), device=torch.device(DEV))
    opt = FusedSGD(flat, lr=0.01, momentum=0.9)
    scaler = GradScaler(init_scale=2.0 ** 12)
    x = torch.randn(8, 3, 64, 64, device=DEV)
    y = torch.randint(0, 10, (8,), device=DEV)
    losses = []
    for _ in range(6):
        opt.zero_grad()
        loss = OF.softmax_cross_entropy(m(x), y)
        scaler.scale(loss).backward()
        skip = scaler.unscale_and_check(flat.grad)
        assert not skip
        opt.step()
        losses.append(float(loss.detach()))
    assert all(l == l for l in losses)
    assert min(losses[1:]) < losses[0], losses
    # inf injection -> skip + backoff
    s0 = scaler.get_scale()
    opt.zero_grad()
    loss = OF.softmax_cross_entropy(m(x), y)
    scaler.scale(loss).backward()
    flat.grad[123] = float("inf")
    assert scaler.unscale_and_check(flat.grad)
    assert scaler.get_scale() == s0 * 0.5


@pytest.mark.gpu
def test_hip_graph_step_matches_eager():
    """Whole-step hipGraph replay must produce the same parameters as the
    same steps run eagerly (same seed, same data)."""
    import torch
    from pytorch_distributed_train_amd.flags import build_parser
    from pytorch_distributed_train_amd.engine import Trainer

    def make():
        args = build_parser().parse_args([
            "--model", "resnet18", "--data", "synthetic-imagenet",
            "--dtype", "bf16", "--batch-size", "8", "--seed", "33"])
        t = Trainer(args)
        t.model.train()
        return t

    g = torch.Generator().manual_seed(17)
    xs = [torch.randn(8, 3, 224, 224, generator=g).cuda() for _ in range(3)]
    ys = [torch.randint(0, 1000, (8,), generator=g).cuda() for _ in range(3)]

    ta = make()
    assert ta.enable_hip_graph(xs[0], ys[0])  # capture (3 warmup steps)
    ta.graph_step(xs[1], ys[1])
    ta.graph_step(xs[2], ys[2])
    torch.cuda.synchronize()

    tb = make()
    for _ in range(3):           # eager twin of the capture warmup
        tb.train_step(xs[0], ys[0])
    tb.train_step(xs[1], ys[1])
    tb.train_step(xs[2], ys[2])
    torch.cuda.synchronize()

    diff = (ta.flat.master - tb.flat.master).abs().max()
    assert torch.allclose(ta.flat.master, tb.flat.master,
                          atol=1e-6, rtol=1e-5), diff


@pytest.mark.gpu
@pytest.mark.parametrize("name", ["resnet34", "resnet101"])
def test_depth_variants_step(name):
    """New depth variants take a full fwd+bwd+step on the HIP path."""
    import torch
    from pytorch_distributed_train_amd.flags import build_parser
    from pytorch_distributed_train_amd.engine import Trainer

    args = build_parser().parse_args([
        "--model", name, "--data", "synthetic-imagenet",
        "--dtype", "bf16", "--batch-size", "4", "--seed", "2",
        "--lr", "0.001"])
    t = Trainer(args)
    t.model.train()
    x = torch.randn(4, 3, 224, 224).cuda()
    y = torch.randint(0, 1000, (4,)).cuda()
    p0 = t.flat.master.clone()
    for _ in range(3):
        loss = t.train_step(x, y)
    assert torch.isfinite(loss), float(loss)
    assert not torch.equal(p0, t.flat.master)  # optimizer stepped


@pytest.mark.gpu
def test_residual_fork_gradients_fused_not_added():
    """Residual-fork gradients are folded into the conv1 bwd-data
    epilogue (GradHolder protocol: producers return None to autograd).
    Oracle: fused vs unfused (PDTA_NO_FORK_FUSE=1) GPU runs differ only
    in the rounding of the fold, and both track a CPU fp32 reference."""
    import os
    import torch
    from pytorch_distributed_train_amd.flags import build_parser
    from pytorch_distributed_train_amd.engine import Trainer

    def make(dev, dtype):
        args = build_parser().parse_args([
            "--model", "resnet50", "--data", "synthetic-imagenet",
            "--dtype", dtype, "--device", dev, "--batch-size", "8",
            "--seed", "41"])
        t = Trainer(args)
        t.model.train()
        return t

    g = torch.Generator().manual_seed(3)
    x = torch.randn(8, 3, 224, 224, generator=g)
    y = torch.randint(0, 1000, (8,), generator=g)

    def grads(dev, dtype, steps=1):
        tr = make(dev, dtype)
        xb, yb = x.to(tr.device), y.to(tr.device)
        for _ in range(steps):
            tr.train_step(xb, yb)
        if dev == "cuda":
            torch.cuda.synchronize()
        return tr.flat.grad.cpu().clone()

    g_fused = grads("cuda", "bf16")
    os.environ["PDTA_NO_FORK_FUSE"] = "1"
    try:
        g_plain = grads("cuda", "bf16")
    finally:
        os.environ.pop("PDTA_NO_FORK_FUSE", None)
    # single step: the only difference is the rounding of the fold
    # (measured whole-flat rel ~0.01-0.02; multi-step comparisons are
    # chaotically unstable through bf16 BN/ReLU and prove nothing)
    rel_ff = float((g_fused - g_plain).norm() /
                   g_plain.norm().clamp(min=1e-12))
    assert rel_ff < 0.05, f"fused vs unfused GPU rel diff {rel_ff}"

    g_cpu = grads("cpu", "fp32")
    rel_f = float((g_fused - g_cpu).norm() / g_cpu.norm().clamp(min=1e-12))
    rel_p = float((g_plain - g_cpu).norm() / g_cpu.norm().clamp(min=1e-12))
    # fused must be as accurate as the plain bf16 path (competitive oracle)
    assert rel_f <= 1.5 * rel_p + 1e-6, (rel_f, rel_p)
