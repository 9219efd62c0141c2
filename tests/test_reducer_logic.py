"""Pure-unit reducer logic (SURVEY.md §4 tier 1-2): bucket assignment over
the flat buffer, unused-parameter flush, and a loss-decreases smoke."""
import torch
import pytest

from pytorch_distributed_train_amd.parallel.flat import FlatParams
from pytorch_distributed_train_amd.parallel.ddp import GradReducer
from pytorch_distributed_train_amd.parallel.comm import Work


class FakeComm:
    rank, world_size = 0, 2

    def __init__(self):
        self.calls = []  # (start-like id, numel) per all-reduce

    def allreduce_avg(self, t, stream=None):
        self.calls.append(t.numel())
        return Work()

    def broadcast(self, t, src=0, stream=None):
        return Work()


def _model():
    torch.manual_seed(0)
    return torch.nn.Sequential(
        torch.nn.Linear(256, 256), torch.nn.ReLU(),
        torch.nn.Linear(256, 256), torch.nn.Linear(256, 10))


def test_bucket_assignment_covers_flat_buffer_exactly():
    m = _model()
    flat = FlatParams(m.named_parameters())
    red = GradReducer(flat, FakeComm(), bucket_cap_mb=1, overlap=True)
    # buckets are contiguous, ordered, non-overlapping, and cover [0, total)
    assert red.buckets[0].start == 0
    assert red.buckets[-1].end == flat.total
    for a, b in zip(red.buckets, red.buckets[1:]):
        assert a.end == b.start
    # every param belongs to exactly one bucket and lies inside its range
    for i in range(flat.num_params):
        b = red._param_bucket[i]
        assert b.start <= flat.offsets[i] < b.end
    # cap respected (except single-param buckets may exceed)
    cap = 1 << 20 // 4
    for b in red.buckets[:-1]:
        assert b.end - b.start >= 1  # non-empty


def test_unused_param_bucket_flushed_at_finalize():
    m = _model()
    flat = FlatParams(m.named_parameters())
    comm = FakeComm()
    red = GradReducer(flat, comm, bucket_cap_mb=1, overlap=True)
    x = torch.randn(4, 256)
    red.prepare()
    # only the FIRST linear participates: later params get no grads
    m[0](x).sum().backward()
    red.finalize()  # must flush every bucket (zeros for unused params)
    assert sum(comm.calls) == flat.total
    assert len(comm.calls) == len(red.buckets)


def test_training_loss_decreases_cpu():
    from pytorch_distributed_train_amd.flags import build_parser
    from pytorch_distributed_train_amd.engine import Trainer

    args = build_parser().parse_args([
        "--model", "resnet18", "--data", "synthetic-cifar10",
        "--dtype", "fp32", "--device", "cpu", "--batch-size", "8",
        "--lr", "0.05", "--seed", "3"])
    t = Trainer(args)
    t.model.train()
    g = torch.Generator().manual_seed(5)
    x = torch.randn(8, 3, 32, 32, generator=g)
    y = torch.randint(0, 10, (8,), generator=g)
    losses = [float(t.train_step(x, y)) for _ in range(8)]
    assert losses[-1] < losses[0] * 0.8, losses  # same batch memorized


def test_param_ready_duplicate_raises():
    """Direct-grad safety: a second arrival for the same param in one
    iteration must raise (a weight-shared model would otherwise train
    on silently-overwritten gradients)."""
    import pytest as _pytest
    import torch as T
    from pytorch_distributed_train_amd.parallel.flat import FlatParams
    from pytorch_distributed_train_amd.parallel.ddp import GradReducer

    class FakeComm:
        rank, world_size = 0, 2

        def allreduce_avg(self, t, stream=None):
            from pytorch_distributed_train_amd.parallel.comm import Work
            return Work()

        def broadcast(self, t, src=0, stream=None):
            from pytorch_distributed_train_amd.parallel.comm import Work
            return Work()

    T.manual_seed(0)
    m = T.nn.Linear(8, 8)
    flat = FlatParams(m.named_parameters())
    red = GradReducer(flat, FakeComm(), bucket_cap_mb=1)
    red.prepare()
    red.param_ready(0)
    with _pytest.raises(RuntimeError, match="arrived twice"):
        red.param_ready(0)


def test_take_over_arrival_removes_hook():
    """After take_over_arrival, only the manual notify counts the param
    (torch fires post-accumulate hooks even for None grads)."""
    import torch as T
    from pytorch_distributed_train_amd.parallel.flat import FlatParams
    from pytorch_distributed_train_amd.parallel.ddp import GradReducer

    class FakeComm:
        rank, world_size = 0, 2
        calls = 0

        def allreduce_avg(self, t, stream=None):
            FakeComm.calls += 1
            from pytorch_distributed_train_amd.parallel.comm import Work
            return Work()

        def broadcast(self, t, src=0, stream=None):
            from pytorch_distributed_train_amd.parallel.comm import Work
            return Work()

    T.manual_seed(0)
    m = T.nn.Linear(8, 8)
    flat = FlatParams(m.named_parameters())
    red = GradReducer(flat, FakeComm(), bucket_cap_mb=1)
    for i in range(flat.num_params):
        red.take_over_arrival(i)
    red.prepare()
    x = T.randn(2, 8)
    m(x).sum().backward()        # hooks gone: no arrivals counted
    assert all(b.pending == len(b.param_ids) for b in red.buckets)
    for i in range(flat.num_params):
        red.param_ready(i)       # manual notifies complete the buckets
    assert all(b.pending == 0 for b in red.buckets)
    red.finalize()
    assert FakeComm.calls >= 1


def test_launch_order_enforced_ascending():
    """An early-ready later bucket is deferred until its predecessors
    launch (identical collective order on every rank)."""
    import torch as T
    from pytorch_distributed_train_amd.parallel.flat import FlatParams
    from pytorch_distributed_train_amd.parallel.ddp import GradReducer

    order = []

    class FakeComm:
        rank, world_size = 0, 2

        def allreduce_avg(self, t, stream=None):
            order.append(t.data_ptr())
            from pytorch_distributed_train_amd.parallel.comm import Work
            return Work()

        def broadcast(self, t, src=0, stream=None):
            from pytorch_distributed_train_amd.parallel.comm import Work
            return Work()

    T.manual_seed(0)
    m = T.nn.Sequential(T.nn.Linear(512, 512), T.nn.Linear(512, 512))
    flat = FlatParams(m.named_parameters())
    red = GradReducer(flat, FakeComm(), bucket_cap_mb=1)
    assert len(red.buckets) >= 2
    red.prepare()
    # complete the LAST bucket first: nothing may launch yet
    last = red.buckets[-1]
    for pid in last.param_ids:
        red.param_ready(pid)
    assert order == [] and last.ready
    # then earlier buckets: everything launches, ascending by start
    for b in red.buckets[:-1]:
        for pid in b.param_ids:
            red.param_ready(pid)
    starts = [red.flat.grad[b.start:b.end].data_ptr()
              for b in red.buckets]
    assert order == starts


def test_ddp_wrapper_api_parity():
    """The torch-style wrapper (model = DDP(module, ...)) drives the
    reducer through plain forward/backward — API parity surface for
    users coming from nn.parallel.DistributedDataParallel."""
    import torch as T
    from pytorch_distributed_train_amd.parallel.flat import FlatParams
    from pytorch_distributed_train_amd.parallel.ddp import (
        DistributedDataParallel)

    class FakeComm:
        rank, world_size = 0, 2
        reduced = 0

        def allreduce_avg(self, t, stream=None):
            FakeComm.reduced += 1
            from pytorch_distributed_train_amd.parallel.comm import Work
            return Work()

        def broadcast(self, t, src=0, stream=None):
            from pytorch_distributed_train_amd.parallel.comm import Work
            return Work()

    T.manual_seed(0)
    m = T.nn.Sequential(T.nn.Linear(16, 16), T.nn.ReLU(),
                        T.nn.Linear(16, 4))
    flat = FlatParams(m.named_parameters())
    ddp = DistributedDataParallel(m, flat, FakeComm(), bucket_cap_mb=1)
    x = T.randn(4, 16)
    out = ddp(x)
    out.sum().backward()
    ddp.reducer.finalize()
    assert FakeComm.reduced >= 1
    assert T.isfinite(flat.grad).all()
