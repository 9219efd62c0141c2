"""Multi-process CPU integration (BASELINE config #1): gloo world_size=2.

Oracle: with SyncBatchNorm, DDP gradients on ranks holding
the two halves of a global batch must equal the single-process gradients
on the full batch; with identical per-rank batches, plain-BN DDP gradients
equal the single-process gradients (all-reduce AVG of identical grads).
"""
import os
import socket

import pytest
import torch
import torch.multiprocessing as mp

from pytorch_distributed_train_amd.flags import build_parser

pytestmark = pytest.mark.timeout(300)


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _make_args(extra=()):
    return build_parser().parse_args([
        "--model", "resnet18", "--data", "synthetic-cifar10",
        "--dtype", "fp32", "--device", "cpu", "--backend", "gloo",
        "--batch-size", "4", "--seed", "11", *extra])


def _global_batch(n=8):
    g = torch.Generator().manual_seed(99)
    x = torch.randn(n, 3, 32, 32, generator=g)
    y = torch.randint(0, 10, (n,), generator=g)
    return x, y


def _worker(rank, world, port, sync_bn, identical_batches, out_dir,
            extra_flags=()):
    os.environ.update(RANK=str(rank), LOCAL_RANK=str(rank),
                      WORLD_SIZE=str(world), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    from pytorch_distributed_train_amd.engine import Trainer
    extra = (["--sync-bn"] if sync_bn else []) + list(extra_flags)
    trainer = Trainer(_make_args(extra))
    x, y = _global_batch()
    if identical_batches:
        xb, yb = x[:4], y[:4]
    else:
        xb, yb = x[rank * 4:(rank + 1) * 4], y[rank * 4:(rank + 1) * 4]
    trainer.model.train()
    loss = trainer.train_step(xb, yb)
    torch.save({"grad": trainer.flat.grad.clone(),
                "master": trainer.flat.master.clone(),
                "loss": loss},
               os.path.join(out_dir, f"rank{rank}.pt"))
    import torch.distributed as dist
    dist.barrier()
    dist.destroy_process_group()


def _single_process_grads(x, y, sync_semantics):
    from pytorch_distributed_train_amd.engine import Trainer
    trainer = Trainer(_make_args())
    trainer.model.train()
    trainer.train_step(x, y)
    return trainer.flat.grad.clone(), trainer.flat.master.clone()


def _run_workers(sync_bn, identical_batches, tmp_path, extra_flags=()):
    port = _free_port()
    ctx = mp.get_context("spawn")
    procs = []
    for r in range(2):
        p = ctx.Process(target=_worker,
                        args=(r, 2, port, sync_bn, identical_batches,
                              str(tmp_path), tuple(extra_flags)))
        p.start()
        procs.append(p)
    for p in procs:
        p.join(240)
        assert p.exitcode == 0, f"worker failed rc={p.exitcode}"
    r0 = torch.load(tmp_path / "rank0.pt", weights_only=False)
    r1 = torch.load(tmp_path / "rank1.pt", weights_only=False)
    return r0, r1


def test_ddp_identical_batches_plain_bn(tmp_path):
    r0, r1 = _run_workers(sync_bn=False, identical_batches=True,
                          tmp_path=tmp_path)
    # identical inputs -> identical grads; AVG all-reduce must be a no-op
    assert torch.allclose(r0["grad"], r1["grad"], atol=1e-6)
    x, y = _global_batch()
    ref_grad, ref_master = _single_process_grads(x[:4], y[:4], False)
    assert torch.allclose(r0["grad"], ref_grad, atol=1e-5), \
        (r0["grad"] - ref_grad).abs().max()
    # one SGD step on identical grads -> identical params across ranks
    assert torch.allclose(r0["master"], r1["master"], atol=1e-6)
    assert torch.allclose(r0["master"], ref_master, atol=1e-5)


def test_ddp_split_batch_syncbn_matches_single_process(tmp_path):
    r0, r1 = _run_workers(sync_bn=True, identical_batches=False,
                          tmp_path=tmp_path)
    assert torch.allclose(r0["grad"], r1["grad"], atol=1e-6)
    x, y = _global_batch()
    ref_grad, _ = _single_process_grads(x, y, True)
    assert torch.allclose(r0["grad"], ref_grad, atol=3e-5), \
        (r0["grad"] - ref_grad).abs().max()


def test_grad_compression_bf16(tmp_path):
    """comm_dtype=bf16 halves all-reduce payload; grads must match the
    uncompressed run to bf16 precision."""
    import torch as T
    from pytorch_distributed_train_amd.parallel.flat import FlatParams
    from pytorch_distributed_train_amd.parallel.ddp import GradReducer

    class FakeComm:
        rank, world_size = 0, 2

        def __init__(self):
            self.payload_dtypes = []

        def allreduce_avg(self, t, stream=None):
            self.payload_dtypes.append(t.dtype)
            t.mul_(1.0)  # pretend-average (identical ranks)
            from pytorch_distributed_train_amd.parallel.comm import Work
            return Work()

        def broadcast(self, t, src=0, stream=None):
            from pytorch_distributed_train_amd.parallel.comm import Work
            return Work()

    T.manual_seed(0)
    m = T.nn.Sequential(T.nn.Linear(64, 64), T.nn.ReLU(),
                        T.nn.Linear(64, 8))
    flat = FlatParams(m.named_parameters())
    comm = FakeComm()
    red = GradReducer(flat, comm, bucket_cap_mb=1, overlap=True,
                      comm_dtype=T.bfloat16)
    x = T.randn(4, 64)
    red.prepare()
    m(x).pow(2).mean().backward()
    ref = flat.grad.clone()
    red.finalize()
    assert all(d == T.bfloat16 for d in comm.payload_dtypes)
    assert len(comm.payload_dtypes) >= 1
    # round-trip through bf16: grads equal to bf16 precision
    assert T.allclose(flat.grad, ref.to(T.bfloat16).float(),
                      atol=0, rtol=0)


def test_ddp_bf16_compressed_comm_end_to_end(tmp_path):
    """--grad-comm-dtype bf16 through the full Trainer: ranks stay
    identical and match the uncompressed gradients to bf16 precision."""
    r0, r1 = _run_workers(sync_bn=False, identical_batches=True,
                          tmp_path=tmp_path,
                          extra_flags=("--grad-comm-dtype", "bf16"))
    assert torch.allclose(r0["grad"], r1["grad"], atol=1e-6)
    assert torch.allclose(r0["master"], r1["master"], atol=1e-6)
    x, y = _global_batch()
    ref_grad, _ = _single_process_grads(x[:4], y[:4], False)
    # identical ranks -> AVG is identity up to the bf16 round-trip
    err = (r0["grad"] - ref_grad).abs()
    tol = ref_grad.abs() * 2 ** -8 + 1e-5
    assert (err <= tol).float().mean() > 0.999, err.max()


def _buf_eval_worker(rank, world, port, out_dir):
    os.environ.update(RANK=str(rank), LOCAL_RANK=str(rank),
                      WORLD_SIZE=str(world), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    from pytorch_distributed_train_amd.engine import Trainer
    from pytorch_distributed_train_amd.flags import DistEnv
    from pytorch_distributed_train_amd.data import make_dataset
    import train as train_mod
    trainer = Trainer(_make_args())
    x, y = _global_batch()
    xb, yb = x[rank * 4:(rank + 1) * 4], y[rank * 4:(rank + 1) * 4]
    trainer.model.train()
    for _ in range(2):  # different data per rank -> running stats diverge
        trainer.train_step(xb, yb)
    trainer._sync_buffers()  # what the NEXT step's pre-forward would do
    bufs = {n: b.clone() for n, b in trainer.model.named_buffers()
            if b.dtype == torch.float32}
    env = DistEnv.from_env()
    ds = make_dataset("synthetic-cifar10", 64, seed=7)
    t1, t5 = train_mod.evaluate(trainer, ds, env, batch_size=8,
                                max_batches=2)
    torch.save({"bufs": bufs, "top1": t1, "top5": t5},
               os.path.join(out_dir, f"be{rank}.pt"))
    import torch.distributed as dist
    dist.barrier()
    dist.destroy_process_group()


def test_buffer_broadcast_and_eval_reduction(tmp_path):
    """--broadcast-buffers (default on): after the pre-forward sync, BN
    running stats are identical across ranks even though per-rank batches
    differed; evaluate() returns the same globally-reduced accuracy on
    every rank (VERDICT r1 missing #3/#4)."""
    port = _free_port()
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_buf_eval_worker,
                         args=(r, 2, port, str(tmp_path)))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(240)
        assert p.exitcode == 0, f"worker failed rc={p.exitcode}"
    r0 = torch.load(tmp_path / "be0.pt", weights_only=False)
    r1 = torch.load(tmp_path / "be1.pt", weights_only=False)
    assert set(r0["bufs"]) == set(r1["bufs"]) and r0["bufs"]
    for n in r0["bufs"]:
        assert torch.equal(r0["bufs"][n], r1["bufs"][n]), n
    assert r0["top1"] == r1["top1"] and r0["top5"] == r1["top5"]


def _ragged_syncbn_worker(rank, world, port, out_dir):
    os.environ.update(RANK=str(rank), LOCAL_RANK=str(rank),
                      WORLD_SIZE=str(world), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    from pytorch_distributed_train_amd.flags import DistEnv
    from pytorch_distributed_train_amd.parallel.comm import make_communicator
    from pytorch_distributed_train_amd.parallel.sync_bn import (
        SyncBatchNorm2dNHWC)
    env = DistEnv.from_env()
    comm = make_communicator(env, "cpu", "gloo")
    C = 8
    g = torch.Generator().manual_seed(5)
    x_full = torch.randn(8, 4, 4, C, generator=g)
    w_full = torch.randn(8, 4, 4, C, generator=g)
    sl = slice(0, 6) if rank == 0 else slice(6, 8)  # RAGGED: 6 vs 2
    bn = SyncBatchNorm2dNHWC(C)
    bn.comm = comm
    bn.train()
    x = x_full[sl].clone().requires_grad_(True)
    y = bn(x)
    (y * w_full[sl]).sum().backward()
    torch.save({"y": y.detach(), "dx": x.grad.clone(),
                "rm": bn.running_mean.clone(),
                "rv": bn.running_var.clone()},
               os.path.join(out_dir, f"rag{rank}.pt"))
    import torch.distributed as dist
    dist.barrier()
    dist.destroy_process_group()


def test_syncbn_ragged_batches_match_single_process(tmp_path):
    """SyncBN with UNEQUAL per-rank batch sizes (6 vs 2): forward outputs,
    input grads and running stats must match a single-process BN over the
    concatenated batch — exercises the reduced-count path that replaced
    the equal-shapes shortcut."""
    from pytorch_distributed_train_amd.models.layers import BatchNorm2dNHWC
    port = _free_port()
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_ragged_syncbn_worker,
                         args=(r, 2, port, str(tmp_path)))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(240)
        assert p.exitcode == 0, f"worker failed rc={p.exitcode}"
    r0 = torch.load(tmp_path / "rag0.pt", weights_only=False)
    r1 = torch.load(tmp_path / "rag1.pt", weights_only=False)
    C = 8
    g = torch.Generator().manual_seed(5)
    x_full = torch.randn(8, 4, 4, C, generator=g)
    w_full = torch.randn(8, 4, 4, C, generator=g)
    bn = BatchNorm2dNHWC(C)
    bn.train()
    x = x_full.clone().requires_grad_(True)
    y = bn(x)
    (y * w_full).sum().backward()
    assert torch.allclose(torch.cat([r0["y"], r1["y"]]), y.detach(),
                          atol=1e-5)
    assert torch.allclose(torch.cat([r0["dx"], r1["dx"]]), x.grad,
                          atol=1e-4), \
        (torch.cat([r0["dx"], r1["dx"]]) - x.grad).abs().max()
    assert torch.allclose(r0["rm"], bn.running_mean, atol=1e-5)
    assert torch.allclose(r0["rv"], bn.running_var, atol=1e-5)
    assert torch.allclose(r0["rm"], r1["rm"], atol=1e-7)


def _debug_comm_worker(rank, world, port, out_dir, mismatch):
    os.environ.update(RANK=str(rank), LOCAL_RANK=str(rank),
                      WORLD_SIZE=str(world), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port),
                      PDTA_DEBUG_COLLECTIVES="1")
    from pytorch_distributed_train_amd.flags import DistEnv
    from pytorch_distributed_train_amd.parallel.comm import (
        make_communicator, DebugComm)
    env = DistEnv.from_env()
    comm = make_communicator(env, torch.device("cpu"), backend="gloo")
    assert isinstance(comm, DebugComm)
    result = {"ok": None, "err": ""}
    try:
        # matching collectives pass verification and compute correctly
        t = torch.full((4,), float(rank + 1))
        comm.allreduce_sum(t).wait()
        assert torch.equal(t, torch.full((4,), 3.0))
        comm.barrier()
        if mismatch == "numel":
            # rank 1 issues a different numel -> BOTH ranks must raise a
            # descriptive error instead of hanging/corrupting
            bad = torch.zeros(8 if rank == 0 else 16)
            comm.allreduce_sum(bad).wait()
        elif mismatch == "op":
            # ranks disagree on WHICH collective comes next — the
            # classic deadlock DETAIL mode exists to catch
            if rank == 0:
                comm.allreduce_sum(torch.zeros(4)).wait()
            else:
                comm.broadcast(torch.zeros(4), src=0).wait()
        result["ok"] = True
    except RuntimeError as e:
        result["ok"] = False
        result["err"] = str(e)
    torch.save(result, os.path.join(out_dir, f"dbg{rank}.pt"))


def test_debug_collectives_cross_check(tmp_path):
    """PDTA_DEBUG_COLLECTIVES=1 (TORCH_DISTRIBUTED_DEBUG=DETAIL analog,
    SURVEY.md §5.2): matched collectives pass; a shape mismatch raises a
    descriptive error on every rank instead of deadlocking."""
    for mismatch in (False, "numel", "op"):
        ctx = mp.get_context("spawn")
        port = _free_port()
        procs = [ctx.Process(target=_debug_comm_worker,
                             args=(r, 2, port, str(tmp_path), mismatch))
                 for r in range(2)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(120)
            assert p.exitcode == 0
        for r in range(2):
            res = torch.load(tmp_path / f"dbg{r}.pt")
            if mismatch:
                assert res["ok"] is False, res
                assert "collective mismatch" in res["err"]
            else:
                assert res["ok"] is True, res
