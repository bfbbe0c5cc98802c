"""Multi-PROCESS GPU integration tests: two ranks share one MI355X via
the gloo backend (CPU-staged p2p), driving the full distributed Worker
+ HIP kernel path — DP hash-sync/serial-equivalence and PP stage
equivalence on real GPU tensors.  (True RCCL runs need one GPU per
rank and are exercised by the driver's multi-GPU scale bench; every
framework code path other than the RCCL transport itself is covered
here.)"""

import os
import random

import pytest
import torch
import torch.multiprocessing as mp

pytestmark = pytest.mark.gpu

SIZES = [48, 32, 24, 16, 12, 10]
GBS, MUB, N, LR = 64, 4, 128, 0.05


def _serial_params_gpu():
    from shallowspeed_amd.data import Dataset
    from shallowspeed_amd.models import MLP, SGD
    from shallowspeed_amd.parallel import NaiveParallelSchedule, Topology, Worker

    dev = torch.device("cuda", 0)
    model = MLP(SIZES, 0, 1, GBS, loss="mse").materialize_device(dev)
    opt = SGD(model.parameters(), lr=LR)
    ds = Dataset(GBS, GBS // MUB, n_samples=N, in_dim=SIZES[0],
                 n_classes=SIZES[-1], device=dev).load(0, 1)
    w = Worker(Topology(device=dev), model, ds, opt)
    for b in range(ds.num_batches()):
        w.execute(NaiveParallelSchedule(ds.num_mubatches(), 1, 0), b)
    torch.cuda.synchronize()
    return [p.data.float().cpu() for p in model.parameters()]


def _entry(rank, world, port, fn_name, out_dir, kwargs):
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK="0")  # both ranks share cuda:0
    import tests.test_gpu_distributed as me

    getattr(me, fn_name)(rank, world, out_dir, **kwargs)


def _run(fn, world, tmp_path, **kwargs):
    mp.spawn(_entry, args=(world, random.randint(20000, 45000), fn.__name__,
                           str(tmp_path), kwargs), nprocs=world, join=True)


def _dp_gpu(rank, world, out_dir):
    from shallowspeed_amd.data import Dataset
    from shallowspeed_amd.models import MLP, SGD
    from shallowspeed_amd.parallel import SCHEDULES, Worker, init_topology
    from shallowspeed_amd.utils import assert_sync, get_model_hash

    dev = torch.device("cuda", 0)
    topo = init_topology(dp=world, pp=1, backend="gloo", device=dev)
    model = MLP(SIZES, 0, 1, GBS, loss="mse").materialize_device(dev)
    opt = SGD(model.parameters(), lr=LR)
    ds = Dataset(GBS, (GBS // world) // MUB, n_samples=N, in_dim=SIZES[0],
                 n_classes=SIZES[-1], device=dev).load(topo.dp_rank, world)
    w = Worker(topo, model, ds, opt)
    for b in range(ds.num_batches()):
        w.execute(SCHEDULES["gpipe"](ds.num_mubatches(), 1, 0), b)
    torch.cuda.synchronize()
    assert_sync(topo.dp_group, get_model_hash(model))
    if rank == 0:
        torch.save([p.data.float().cpu() for p in model.parameters()],
                   os.path.join(out_dir, "dp_gpu.pt"))
    torch.distributed.destroy_process_group()


def _pp_grads(rank, world, out_dir, schedule="gpipe"):
    """One batch, lr=0: stage grads must match serial grads tightly —
    catches buffer-aliasing bugs that loose weight tolerances hide."""
    from shallowspeed_amd.data import Dataset
    from shallowspeed_amd.models import MLP, SGD
    from shallowspeed_amd.parallel import SCHEDULES, Worker, init_topology

    dev = torch.device("cuda", 0)
    topo = init_topology(dp=1, pp=world, backend="gloo", device=dev)
    model = MLP(SIZES, topo.stage_id, world, GBS,
                loss="mse").materialize_device(dev)
    opt = SGD(model.parameters(), lr=0.0)
    ds = Dataset(GBS, GBS // MUB, n_samples=N, in_dim=SIZES[0],
                 n_classes=SIZES[-1], device=dev).load(0, 1)
    w = Worker(topo, model, ds, opt)
    w.execute(SCHEDULES[schedule](ds.num_mubatches(), world,
                                  topo.stage_id), 0)
    torch.cuda.synchronize()
    torch.save([p.grad.float().cpu() for p in model.parameters()],
               os.path.join(out_dir, f"ppg_{topo.stage_id}.pt"))
    torch.distributed.destroy_process_group()


@pytest.mark.parametrize("schedule", ["gpipe", "pipedream", "naive"])
def test_pp2_grads_match_serial_tight(tmp_path, gpu_device, schedule):
    from shallowspeed_amd.data import Dataset
    from shallowspeed_amd.models import MLP, SGD
    from shallowspeed_amd.parallel import NaiveParallelSchedule, Topology, Worker

    _run(_pp_grads, 2, tmp_path, schedule=schedule)
    got = torch.load(tmp_path / "ppg_0.pt", weights_only=False) + \
        torch.load(tmp_path / "ppg_1.pt", weights_only=False)

    dev = torch.device("cuda", 0)
    model = MLP(SIZES, 0, 1, GBS, loss="mse").materialize_device(dev)
    opt = SGD(model.parameters(), lr=0.0)
    ds = Dataset(GBS, GBS // MUB, n_samples=N, in_dim=SIZES[0],
                 n_classes=SIZES[-1], device=dev).load(0, 1)
    w = Worker(Topology(device=dev), model, ds, opt)
    w.execute(NaiveParallelSchedule(ds.num_mubatches(), 1, 0), 0)
    torch.cuda.synchronize()
    want = [p.grad.float().cpu() for p in model.parameters()]
    assert len(got) == len(want)
    for g, r in zip(got, want):
        torch.testing.assert_close(g, r, atol=2e-3, rtol=1e-2)


def _pp_gpu(rank, world, out_dir, schedule="pipedream"):
    from shallowspeed_amd.data import Dataset
    from shallowspeed_amd.models import MLP, SGD
    from shallowspeed_amd.parallel import SCHEDULES, Worker, init_topology

    dev = torch.device("cuda", 0)
    topo = init_topology(dp=1, pp=world, backend="gloo", device=dev)
    model = MLP(SIZES, topo.stage_id, world, GBS,
                loss="mse").materialize_device(dev)
    opt = SGD(model.parameters(), lr=LR)
    ds = Dataset(GBS, GBS // MUB, n_samples=N, in_dim=SIZES[0],
                 n_classes=SIZES[-1], device=dev).load(0, 1)
    w = Worker(topo, model, ds, opt)
    for b in range(ds.num_batches()):
        w.execute(SCHEDULES[schedule](ds.num_mubatches(), world,
                                      topo.stage_id), b)
    torch.cuda.synchronize()
    torch.save([p.data.float().cpu() for p in model.parameters()],
               os.path.join(out_dir, f"pp_gpu_{topo.stage_id}.pt"))
    torch.distributed.destroy_process_group()


def test_dp2_on_gpu_matches_serial(tmp_path, gpu_device):
    _run(_dp_gpu, 2, tmp_path)
    got = torch.load(tmp_path / "dp_gpu.pt", weights_only=False)
    want = _serial_params_gpu()
    for g, w in zip(got, want):
        # bf16 compute + atomic-order jitter: tolerance-based equality
        torch.testing.assert_close(g, w, atol=5e-3, rtol=5e-2)


@pytest.mark.parametrize("schedule", ["gpipe", "pipedream"])
def test_pp2_on_gpu_matches_serial(tmp_path, gpu_device, schedule):
    _run(_pp_gpu, 2, tmp_path, schedule=schedule)
    got = torch.load(tmp_path / "pp_gpu_0.pt", weights_only=False) + \
        torch.load(tmp_path / "pp_gpu_1.pt", weights_only=False)
    want = _serial_params_gpu()
    assert len(got) == len(want)
    for g, w in zip(got, want):
        torch.testing.assert_close(g, w, atol=5e-3, rtol=5e-2)


def _dp_gpu_det(rank, world, out_dir, tag="a"):
    """DP training with --deterministic semantics: split_k=1 wgrad +
    fixed bucket order.  Saves the final model hash for bitwise
    comparison across runs."""
    from shallowspeed_amd.data import Dataset
    from shallowspeed_amd.models import MLP, SGD
    from shallowspeed_amd.ops import functional as F
    from shallowspeed_amd.parallel import SCHEDULES, Worker, init_topology
    from shallowspeed_amd.utils import assert_sync, get_model_hash

    F.set_deterministic(True)
    dev = torch.device("cuda", 0)
    topo = init_topology(dp=world, pp=1, backend="gloo", device=dev)
    model = MLP(SIZES, 0, 1, GBS, loss="mse").materialize_device(dev)
    opt = SGD(model.parameters(), lr=LR)
    ds = Dataset(GBS, (GBS // world) // MUB, n_samples=N, in_dim=SIZES[0],
                 n_classes=SIZES[-1], device=dev).load(topo.dp_rank, world)
    w = Worker(topo, model, ds, opt)
    for b in range(ds.num_batches()):
        w.execute(SCHEDULES["gpipe"](ds.num_mubatches(), 1, 0), b)
    torch.cuda.synchronize()
    # bitwise replica equality (the reference's exact-equality gate,
    # train.py:154-155, at full strength on GPU)
    assert_sync(topo.dp_group, get_model_hash(model))
    if rank == 0:
        with open(os.path.join(out_dir, f"det_hash_{tag}"), "w") as f:
            f.write(get_model_hash(model))
    torch.distributed.destroy_process_group()


def test_deterministic_dp2_bitwise_reproducible(tmp_path, gpu_device):
    """--deterministic mode: two identical DP=2 trainings produce
    BITWISE-identical models (sha1 hash equality), and replicas are
    bitwise in sync — no 5e-3 tolerances."""
    _run(_dp_gpu_det, 2, tmp_path, tag="a")
    _run(_dp_gpu_det, 2, tmp_path, tag="b")
    ha = (tmp_path / "det_hash_a").read_text()
    hb = (tmp_path / "det_hash_b").read_text()
    assert ha == hb, (ha, hb)


def _pp_mem(rank, world, out_dir, schedule="gpipe"):
    from shallowspeed_amd.data import Dataset
    from shallowspeed_amd.models import MLP, SGD
    from shallowspeed_amd.parallel import SCHEDULES, Worker, init_topology

    dev = torch.device("cuda", 0)
    topo = init_topology(dp=1, pp=world, backend="gloo", device=dev)
    sizes = [1024] * 8 + [10, 10]
    model = MLP(sizes, topo.stage_id, world, 4096,
                loss="mse").materialize_device(dev)
    opt = SGD(model.parameters(), lr=0.01)
    ds = Dataset(4096, 128, n_samples=8192, in_dim=sizes[0],
                 n_classes=sizes[-1], device=dev).load(0, 1)
    w = Worker(topo, model, ds, opt)
    torch.cuda.reset_peak_memory_stats()
    base = torch.cuda.memory_allocated()
    w.execute(SCHEDULES[schedule](ds.num_mubatches(), world,
                                  topo.stage_id), 0)
    torch.cuda.synchronize()
    peak = torch.cuda.max_memory_allocated() - base
    with open(os.path.join(out_dir, f"mem_{schedule}_{topo.stage_id}"),
              "w") as f:
        f.write(str(peak))
    torch.distributed.destroy_process_group()


def test_rccl_world1_smoke(tmp_path, gpu_device):
    """First-contact RCCL smoke on a single GPU: init_process_group
    with backend nccl (== RCCL on ROCm), device_id-bound eager comm
    creation, a device all_reduce, a device-bound barrier, and clean
    destroy — run in a subprocess so the nccl state can't leak into
    other tests.  (True multi-GPU RCCL needs one GPU per rank and is
    exercised by the driver's scale run; this validates the init path
    that run will take.)"""
    import subprocess
    import sys

    code = r"""
import os, datetime, torch, torch.distributed as dist
os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT="29541",
                  RANK="0", WORLD_SIZE="1", LOCAL_RANK="0")
dev = torch.device("cuda", 0)
torch.cuda.set_device(dev)
dist.init_process_group("nccl", device_id=dev,
                        timeout=datetime.timedelta(seconds=60))
t = torch.ones(1 << 20, device=dev)
dist.all_reduce(t)
dist.barrier(device_ids=[0])
torch.cuda.synchronize()
assert t.sum().item() == float(1 << 20)
dist.destroy_process_group()
print("RCCL_SMOKE_OK")
"""
    r = subprocess.run([sys.executable, "-c", code], capture_output=True,
                       text=True, timeout=300)
    assert r.returncode == 0 and "RCCL_SMOKE_OK" in r.stdout, \
        r.stdout + r.stderr


def test_real_digits_convergence_gpu(tmp_path, gpu_device):
    """The reference's actual task, end to end on the GPU: real
    handwritten digits (MNIST-shaped), full CLI, bf16 HIP kernels,
    val_acc >= 0.95 (reference trains real MNIST to high accuracy,
    train.py:148-152)."""
    import re
    import subprocess
    import sys

    root = __file__.rsplit("/tests/", 1)[0]
    d = str(tmp_path / "digits")
    r = subprocess.run([sys.executable, "prepare_data.py", "--out", d,
                        "--source", "digits"], cwd=root,
                       capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stdout + r.stderr
    r = subprocess.run([sys.executable, "train.py", "--data-dir", d,
                        "--epochs", "40", "--loss", "xent", "--lr", "0.05",
                        "--momentum", "0.9", "--schedule", "gpipe",
                        "--mubatches", "4", "--device", "cuda"],
                       cwd=root, capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, r.stdout + r.stderr
    accs = [float(m) for m in re.findall(r"val_acc=([0-9.]+)", r.stdout)]
    assert accs and max(accs) >= 0.95, r.stdout[-2000:]


def test_1f1b_bounds_activation_memory(tmp_path, gpu_device):
    """PipeDream-Flush's raison d'être: stage-0 peak activation memory
    is bounded by warmup+1 µbatches, while GPipe holds all M.  With 16
    µbatches and 2 stages the stage-0 peak must be several times
    smaller under 1F1B (this also guards the deferred-wgrad window —
    unbounded deferral would silently destroy the bound; the window
    retains at most 4 extra µbatches)."""
    _run(_pp_mem, 2, tmp_path, schedule="gpipe")
    gpipe = int((tmp_path / "mem_gpipe_0").read_text())
    _run(_pp_mem, 2, tmp_path, schedule="pipedream")
    flush = int((tmp_path / "mem_pipedream_0").read_text())
    assert flush * 3 < gpipe, (flush, gpipe)
