"""Stage-sharded checkpoint save/load/resume."""

import torch

from shallowspeed_amd.checkpoint import load_checkpoint, save_checkpoint
from shallowspeed_amd.data import Dataset
from shallowspeed_amd.models import MLP, SGD
from shallowspeed_amd.parallel import NaiveParallelSchedule, Topology, Worker

SIZES = [24, 16, 12, 10]


def _train(model, steps=2, gbs=16):
    opt = SGD(model.parameters(), lr=0.05)
    ds = Dataset(gbs, 8, n_samples=64, in_dim=SIZES[0],
                 n_classes=SIZES[-1]).load(0, 1)
    w = Worker(Topology(), model, ds, opt)
    for b in range(steps):
        w.execute(NaiveParallelSchedule(ds.num_mubatches(), 1, 0), b)
    return w, ds


def test_roundtrip(tmp_path):
    model = MLP(SIZES, 0, 1, 16).materialize_device("cpu")
    _train(model)
    topo = Topology()
    save_checkpoint(tmp_path, model, topo, step=2, extra={"note": "t"})

    fresh = MLP(SIZES, 0, 1, 16).materialize_device("cpu")
    assert any((a.data != b.data).any()
               for a, b in zip(model.parameters(), fresh.parameters()))
    meta = load_checkpoint(tmp_path, fresh, topo)
    assert meta["step"] == 2 and meta["note"] == "t"
    for a, b in zip(model.parameters(), fresh.parameters()):
        torch.testing.assert_close(a.data, b.data, rtol=0, atol=0)


def test_stage_sharded_layout(tmp_path):
    """Each pipeline stage writes its own file (BASELINE.json
    checkpoint-layout requirement)."""
    import os

    for s in range(2):
        topo = Topology(rank=s, world=2, dp=1, pp=2)
        model = MLP(SIZES, s, 2, 16).materialize_device("cpu")
        save_checkpoint(tmp_path, model, topo, step=1)
    files = sorted(os.listdir(tmp_path))
    assert "stage_00.pt" in files and "stage_01.pt" in files
    # reload each shard into the matching stage
    for s in range(2):
        topo = Topology(rank=s, world=2, dp=1, pp=2)
        m = MLP(SIZES, s, 2, 16).materialize_device("cpu")
        load_checkpoint(tmp_path, m, topo)


def test_resume_continues_identically(tmp_path):
    """train A->save->load->train B  ==  train A+B straight through."""
    straight = MLP(SIZES, 0, 1, 16).materialize_device("cpu")
    _train(straight, steps=4)

    part1 = MLP(SIZES, 0, 1, 16).materialize_device("cpu")
    w, ds = _train(part1, steps=2)
    save_checkpoint(tmp_path, part1, Topology(), step=2)

    part2 = MLP(SIZES, 0, 1, 16).materialize_device("cpu")
    load_checkpoint(tmp_path, part2, Topology())
    opt = SGD(part2.parameters(), lr=0.05)
    w2 = Worker(Topology(), part2, ds, opt)
    for b in (2, 3):
        w2.execute(NaiveParallelSchedule(ds.num_mubatches(), 1, 0), b)

    for a, b in zip(straight.parameters(), part2.parameters()):
        torch.testing.assert_close(a.data, b.data, rtol=0, atol=0)


def _train_momentum(model, opt, ds, batches):
    w = Worker(Topology(), model, ds, opt)
    for b in batches:
        w.execute(NaiveParallelSchedule(ds.num_mubatches(), 1, 0), b)


def test_resume_with_momentum_restores_velocity(tmp_path):
    """Momentum velocities are part of the checkpoint: resuming a
    momentum>0 run continues the exact optimization trajectory
    (round-1 gap: velocities were silently reset on resume)."""
    mk_ds = lambda: Dataset(16, 8, n_samples=64, in_dim=SIZES[0],
                            n_classes=SIZES[-1]).load(0, 1)

    straight = MLP(SIZES, 0, 1, 16).materialize_device("cpu")
    opt_s = SGD(straight.parameters(), lr=0.05, momentum=0.9)
    _train_momentum(straight, opt_s, mk_ds(), [0, 1, 2, 3])

    part1 = MLP(SIZES, 0, 1, 16).materialize_device("cpu")
    opt1 = SGD(part1.parameters(), lr=0.05, momentum=0.9)
    ds = mk_ds()
    _train_momentum(part1, opt1, ds, [0, 1])
    assert any(v.abs().sum() > 0 for v in opt1._vel)
    save_checkpoint(tmp_path, part1, Topology(), step=2, optimizer=opt1)

    part2 = MLP(SIZES, 0, 1, 16).materialize_device("cpu")
    opt2 = SGD(part2.parameters(), lr=0.05, momentum=0.9)
    load_checkpoint(tmp_path, part2, Topology(), optimizer=opt2)
    for a, b in zip(opt1._vel, opt2._vel):
        torch.testing.assert_close(a, b, rtol=0, atol=0)
    _train_momentum(part2, opt2, ds, [2, 3])

    for a, b in zip(straight.parameters(), part2.parameters()):
        torch.testing.assert_close(a.data, b.data, rtol=0, atol=0)


def test_load_rejects_mismatched_pp(tmp_path):
    model = MLP(SIZES, 0, 1, 16).materialize_device("cpu")
    save_checkpoint(tmp_path, model, Topology(), step=1)
    import pytest

    bad_topo = Topology(rank=0, world=2, dp=1, pp=2)
    stage0 = MLP([SIZES[0], SIZES[1]], 0, 1, 16).materialize_device("cpu")
    with pytest.raises(AssertionError, match="pp="):
        load_checkpoint(tmp_path, stage0, bad_topo)
