"""Single-process end-to-end: Worker + schedules train a model on CPU.

Covers the reference's serial-equivalence design property: any
schedule at pp=1 must produce the same weights as plain µbatch-
accumulated training, and training must actually learn (reference
gates: accuracy climbs, train.py:132-152)."""

import pytest
import torch

from shallowspeed_amd.data import Dataset
from shallowspeed_amd.models import MLP, SGD
from shallowspeed_amd.parallel import (
    GPipeSchedule,
    NaiveParallelSchedule,
    PipeDreamFlushSchedule,
    Topology,
    Worker,
)

SIZES = [20, 16, 12, 10]


def make_worker(loss="mse", gbs=32, mub=8, n=64, seed=7):
    topo = Topology()
    model = MLP(SIZES, 0, 1, gbs, loss=loss).materialize_device("cpu")
    opt = SGD(model.parameters(), lr=0.05)
    ds = Dataset(gbs, mub, n_samples=n, in_dim=SIZES[0],
                 n_classes=SIZES[-1], seed=seed)
    ds.load(0, 1)
    return Worker(topo, model, ds, opt), model, ds


@pytest.mark.parametrize("cls", [NaiveParallelSchedule, GPipeSchedule,
                                 PipeDreamFlushSchedule])
def test_schedules_equal_weights_at_pp1(cls):
    """All schedules at pp=1 are gradient-accumulation reorderings of
    the same math → identical weights (f32, same op order per layer)."""
    ref_worker, ref_model, ds = make_worker()
    for b in range(ds.num_batches()):
        ref_worker.execute(NaiveParallelSchedule(ds.num_mubatches(), 1, 0), b)
    w, model, ds2 = make_worker()
    for b in range(ds2.num_batches()):
        w.execute(cls(ds2.num_mubatches(), 1, 0), b)
    for a, b_ in zip(ref_model.parameters(), model.parameters()):
        torch.testing.assert_close(a.data, b_.data)


@pytest.mark.parametrize("loss", ["mse", "xent"])
def test_training_learns(loss):
    w, model, ds = make_worker(loss=loss, gbs=32, mub=8, n=256)
    x, y = ds.x, ds.y

    def acc():
        model.eval()
        probs = model.forward(x, 0)
        model.train()
        return (probs.argmax(-1) == y.argmax(-1)).float().mean().item()

    a0 = acc()
    for epoch in range(30 if loss == "xent" else 60):
        for b in range(ds.num_batches()):
            w.execute(GPipeSchedule(ds.num_mubatches(), 1, 0), b)
    a1 = acc()
    # xent converges fast; the softmax+MSE head (reference parity) has
    # much weaker gradients at equal lr — require solid improvement for
    # both and a strong threshold for xent.
    assert a1 > a0 + 0.2, (a0, a1)
    if loss == "xent":
        assert a1 > 0.5, (a0, a1)


def test_worker_instruction_timing():
    w, model, ds = make_worker()
    w.enable_instruction_timing(True)
    w.execute(NaiveParallelSchedule(ds.num_mubatches(), 1, 0), 0)
    assert "Forward" in w.instruction_times
    assert "OptimizerStep" in w.instruction_times


def test_persistent_buffers_reused():
    w, model, ds = make_worker()
    w.execute(NaiveParallelSchedule(ds.num_mubatches(), 1, 0), 0)
    bufs = (w._in_bufs[0].data_ptr(), w._out_bufs[0].data_ptr())
    w.execute(NaiveParallelSchedule(ds.num_mubatches(), 1, 0), 1)
    assert (w._in_bufs[0].data_ptr(), w._out_bufs[0].data_ptr()) == bufs
