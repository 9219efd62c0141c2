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
