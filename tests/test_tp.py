"""Tensor parallelism (beyond-reference): column/row-parallel Linear
pair equivalence vs the serial model — single-process (tp_world=1
degenerates to serial math) and multi-process (gloo, world 2/4)."""

import os
import random

import pytest
import torch
import torch.multiprocessing as mp

from shallowspeed_amd.data import Dataset
from shallowspeed_amd.models import MLP, SGD
from shallowspeed_amd.parallel import TPMLP, NaiveParallelSchedule, Topology, Worker

SIZES = [24, 32, 16, 10]  # col(24->32) row(32->16) + replicated head
GBS, N, LR = 16, 64, 0.05


def _serial(loss="xent", steps=3):
    model = MLP(SIZES, 0, 1, GBS, loss=loss).materialize_device("cpu")
    opt = SGD(model.parameters(), lr=LR)
    ds = Dataset(GBS, GBS, n_samples=N, in_dim=SIZES[0],
                 n_classes=SIZES[-1]).load(0, 1)
    w = Worker(Topology(), model, ds, opt)
    for b in range(min(steps, ds.num_batches())):
        w.execute(NaiveParallelSchedule(1, 1, 0), b)
    model.eval()
    g = torch.Generator().manual_seed(42)
    xin = torch.randn(8, SIZES[0], generator=g)
    return model.forward(xin), xin


def test_tp1_equals_serial_mlp():
    """tp_world=1: the TP model is numerically the serial model."""
    want, xin = _serial(steps=0)
    tp = TPMLP(SIZES, None, 0, 1, GBS).materialize_device("cpu")
    tp.eval()
    torch.testing.assert_close(tp.forward(xin), want,
                               rtol=1e-5, atol=1e-6)


def _tp_entry(rank, world, port, out_dir):
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    import torch.distributed as dist

    dist.init_process_group("gloo")
    group = dist.group.WORLD
    model = TPMLP(SIZES, group, rank, world, GBS).materialize_device("cpu")
    opt = SGD(model.parameters(), lr=LR)
    ds = Dataset(GBS, GBS, n_samples=N, in_dim=SIZES[0],
                 n_classes=SIZES[-1]).load(0, 1)
    w = Worker(Topology(rank=rank, world=world), model, ds, opt,
               use_dp=False)
    for b in range(3):
        w.execute(NaiveParallelSchedule(1, 1, 0), b)
    model.eval()
    g = torch.Generator().manual_seed(42)
    xin = torch.randn(8, SIZES[0], generator=g)
    out = model.forward(xin)
    if rank == 0:
        torch.save(out, os.path.join(out_dir, "tp_out.pt"))
    dist.destroy_process_group()


@pytest.mark.parametrize("world", [2, 4])
def test_tp_training_matches_serial(tmp_path, world):
    """TP=2/4 training (3 steps) produces the same function as serial
    training: forward on a fixed probe input matches within f32
    reduction-order tolerance."""
    port = random.randint(20000, 45000)
    mp.spawn(_tp_entry, args=(world, port, str(tmp_path)), nprocs=world,
             join=True)
    got = torch.load(tmp_path / "tp_out.pt", weights_only=False)
    want, _ = _serial(steps=3)
    torch.testing.assert_close(got, want, rtol=1e-4, atol=1e-5)


def _tp_gpu_entry(rank, world, port, out_dir):
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world), LOCAL_RANK="0")
    import torch.distributed as dist

    dist.init_process_group("gloo")
    dev = torch.device("cuda", 0)
    group = dist.group.WORLD
    model = TPMLP(SIZES, group, rank, world, GBS).materialize_device(dev)
    opt = SGD(model.parameters(), lr=LR)
    ds = Dataset(GBS, GBS, n_samples=N, in_dim=SIZES[0],
                 n_classes=SIZES[-1], device=dev).load(0, 1)
    w = Worker(Topology(rank=rank, world=world, device=dev), model, ds, opt,
               use_dp=False)
    for b in range(3):
        w.execute(NaiveParallelSchedule(1, 1, 0), b)
    model.eval()
    g = torch.Generator().manual_seed(42)
    xin = torch.randn(8, SIZES[0], generator=g).bfloat16().to(dev)
    out = model.forward(xin)
    torch.cuda.synchronize()
    if rank == 0:
        torch.save(out.float().cpu(), os.path.join(out_dir, "tp_gpu.pt"))
    dist.destroy_process_group()


@pytest.mark.gpu
def test_tp2_gpu_matches_serial(tmp_path, gpu_device):
    """TP=2 on GPU (two ranks share one MI355X via gloo staging, HIP
    kernels + bf16 compute): training matches the serial GPU model."""
    port = random.randint(20000, 45000)
    mp.spawn(_tp_gpu_entry, args=(2, port, str(tmp_path)), nprocs=2,
             join=True)
    got = torch.load(tmp_path / "tp_gpu.pt", weights_only=False)

    dev = torch.device("cuda", 0)
    model = MLP(SIZES, 0, 1, GBS, loss="xent").materialize_device(dev)
    opt = SGD(model.parameters(), lr=LR)
    ds = Dataset(GBS, GBS, n_samples=N, in_dim=SIZES[0],
                 n_classes=SIZES[-1], device=dev).load(0, 1)
    w = Worker(Topology(device=dev), model, ds, opt)
    for b in range(3):
        w.execute(NaiveParallelSchedule(1, 1, 0), b)
    model.eval()
    g = torch.Generator().manual_seed(42)
    xin = torch.randn(8, SIZES[0], generator=g).bfloat16().to(dev)
    want = model.forward(xin).float().cpu()
    torch.cuda.synchronize()
    torch.testing.assert_close(got, want, atol=5e-3, rtol=5e-2)


def _dptp_entry(rank, world, port, out_dir, dp, tp):
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    from shallowspeed_amd.parallel import init_topology
    from shallowspeed_amd.utils import assert_sync, get_model_hash

    topo = init_topology(dp=dp, pp=1, backend="gloo",
                         device=torch.device("cpu"), tp=tp)
    model = TPMLP(SIZES, topo.tp_group, topo.tp_rank, tp,
                  GBS).materialize_device("cpu")
    opt = SGD(model.parameters(), lr=LR)
    ds = Dataset(GBS, GBS // dp, n_samples=N, in_dim=SIZES[0],
                 n_classes=SIZES[-1]).load(topo.dp_rank, dp)
    w = Worker(topo, model, ds, opt)
    for b in range(ds.num_batches()):
        w.execute(NaiveParallelSchedule(1, 1, 0), b)
    # DP replicas of the SAME shard must be hash-identical
    assert_sync(topo.dp_group, get_model_hash(model))
    model.eval()
    g = torch.Generator().manual_seed(42)
    xin = torch.randn(8, SIZES[0], generator=g)
    out = model.forward(xin)
    if rank == 0:
        torch.save(out, os.path.join(out_dir, "dptp_out.pt"))
    import torch.distributed as dist

    dist.destroy_process_group()


def test_dp2_tp2_matches_serial(tmp_path):
    """The composed DP×TP grid (world 4, TP innermost): per-shard DP
    hash sync + the trained function matches serial training."""
    port = random.randint(20000, 45000)
    mp.spawn(_dptp_entry, args=(4, port, str(tmp_path), 2, 2), nprocs=4,
             join=True)
    got = torch.load(tmp_path / "dptp_out.pt", weights_only=False)
    want, _ = _serial(steps=4)
    torch.testing.assert_close(got, want, rtol=1e-4, atol=1e-5)


def _tp_ckpt_entry(rank, world, port, out_dir):
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    from shallowspeed_amd.checkpoint import load_checkpoint, save_checkpoint
    from shallowspeed_amd.parallel import init_topology

    topo = init_topology(dp=1, pp=1, backend="gloo",
                         device=torch.device("cpu"), tp=world)
    model = TPMLP(SIZES, topo.tp_group, topo.tp_rank, world,
                  GBS).materialize_device("cpu")
    opt = SGD(model.parameters(), lr=LR)
    ds = Dataset(GBS, GBS, n_samples=N, in_dim=SIZES[0],
                 n_classes=SIZES[-1]).load(0, 1)
    w = Worker(topo, model, ds, opt, use_dp=False)
    for b in range(2):
        w.execute(NaiveParallelSchedule(1, 1, 0), b)
    save_checkpoint(os.path.join(out_dir, "ck"), model, topo, step=2)
    import torch.distributed as dist

    dist.barrier()  # loaders must not race rank 0's meta.pt write

    fresh = TPMLP(SIZES, topo.tp_group, topo.tp_rank, world,
                  GBS).materialize_device("cpu")
    load_checkpoint(os.path.join(out_dir, "ck"), fresh, topo)
    for a, b2 in zip(model.parameters(), fresh.parameters()):
        torch.testing.assert_close(a.data, b2.data, rtol=0, atol=0)
    import torch.distributed as dist

    dist.destroy_process_group()


def test_tp_checkpoint_shards_per_rank(tmp_path):
    """TP checkpoints write one shard file PER TP RANK (no stage_00
    collision) and round-trip exactly; tp mismatch on load is
    rejected."""
    port = random.randint(20000, 45000)
    mp.spawn(_tp_ckpt_entry, args=(2, port, str(tmp_path)), nprocs=2,
             join=True)
    files = sorted(os.listdir(tmp_path / "ck"))
    assert "stage_00_tp00.pt" in files and "stage_00_tp01.pt" in files
    # tp-mismatch rejection (serial topology reading a tp=2 checkpoint)
    from shallowspeed_amd.checkpoint import load_checkpoint

    m = TPMLP(SIZES, None, 0, 1, GBS).materialize_device("cpu")
    with pytest.raises(AssertionError, match="tp="):
        load_checkpoint(str(tmp_path / "ck"), m, Topology())
