"""Benchmark harness (driver contract).

`python bench.py --gpus N --steps K --warmup W` runs the flagship
training step — 4-layer MLP, MNIST shape (784→256→256→256→10), bf16,
synthetic data, fused softmax-cross-entropy head — on N GPUs of one
node as DP over RCCL/xGMI (weak scaling: fixed per-GPU batch).

For N>1 the driver launches this via torch.distributed.run with one
rank per GPU; RANK/LOCAL_RANK/WORLD_SIZE/MASTER_* come from the env.

Rank 0 prints ONE JSON line with the whole-job samples/sec.  (Under
the production nccl/RCCL backend stdout carries only that line; the
gloo TEST backend prints its own "[Gloo] Rank..." connectivity banners
to stdout — pipe through `grep '^{'` when parsing gloo runs.)
"""

import argparse
import json
import os
import time

import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=50)
    ap.add_argument("--warmup", type=int, default=10)
    ap.add_argument("--local-batch", type=int, default=16384)
    ap.add_argument("--mubatches", type=int, default=1)
    ap.add_argument("--pp", type=int, default=1)
    ap.add_argument("--tp", type=int, default=1,
                    help="tensor parallelism (composes with --dp; "
                         "pp must be 1)")
    ap.add_argument("--schedule", default="naive",
                    choices=["naive", "gpipe", "pipedream"])
    ap.add_argument("--layer-sizes", default="784,256,256,256,10")
    ap.add_argument("--loss", default="xent", choices=["xent", "mse"])
    ap.add_argument("--backend", default=None,
                    help="torch.distributed backend override "
                         "(default: nccl on GPU, gloo on CPU)")
    ap.add_argument("--device", default=None)
    ap.add_argument("--graph", action="store_true",
                    help="capture steps in a hipGraph and replay "
                         "(single-GPU; replay floor ~10µs makes eager "
                         "faster at the flagship step size — measured)")
    ap.add_argument("--no-graph", action="store_true",
                    help="(compat) force eager; eager is the default")
    ap.add_argument("--deterministic", action="store_true",
                    help="bitwise-reproducible mode (split_k=1 wgrad)")
    ap.add_argument("--fp8-fwd", action="store_true",
                    help="MX-fp8 training forward on qualifying layers")
    args = ap.parse_args()

    from shallowspeed_amd.data import Dataset
    from shallowspeed_amd.models import MLP, SGD
    from shallowspeed_amd.parallel import SCHEDULES, Worker, init_topology

    if args.deterministic:
        from shallowspeed_amd.ops import functional as _F

        _F.set_deterministic(True)

    world = int(os.environ.get("WORLD_SIZE", "1"))
    assert world == args.gpus or args.gpus == 1 or world == 1, \
        f"--gpus {args.gpus} vs WORLD_SIZE {world}"
    n = max(world, 1)
    pp = args.pp
    tp = args.tp
    if tp > 1:
        assert pp == 1, "tp composes with dp only"
        dp = n // tp
        assert dp * tp == n, f"dp({dp})*tp({tp}) != {n}"
    else:
        dp = n // pp
        assert dp * pp == n, f"dp({dp})*pp({pp}) != {n}"

    device = torch.device(args.device) if args.device else None
    topo = init_topology(dp, pp, backend=args.backend, device=device,
                         tp=tp)
    device = topo.device
    is_gpu = device.type == "cuda"

    sizes = [int(s) for s in args.layer_sizes.split(",")]
    assert args.local_batch % args.mubatches == 0
    # PP needs len(sizes) % pp == 0 (stage slicing, layers.py:242):
    # pad the default config with extra hidden layers if needed
    while len(sizes) % pp != 0:
        sizes.insert(1, sizes[1])
    assert len(sizes) % pp == 0
    global_batch = args.local_batch * dp
    mubatch = args.local_batch // args.mubatches

    if tp > 1:
        from shallowspeed_amd.parallel import TPMLP

        model = TPMLP(sizes, topo.tp_group, topo.tp_rank, tp, global_batch,
                      loss=args.loss)
    else:
        model = MLP(sizes, topo.stage_id, pp, global_batch, loss=args.loss)
    model.materialize_device(device)
    if args.fp8_fwd:
        model.set_fp8_fwd(True)
    opt = SGD(model.parameters(), lr=0.001)
    # synthetic data of the benchmark shape, random-init weights (no
    # network for datasets/checkpoints in this environment)
    n_samples = max(args.local_batch * dp * 2, 2 * global_batch)
    ds = Dataset(global_batch, mubatch, n_samples=n_samples,
                 in_dim=sizes[0], n_classes=sizes[-1], device=device)
    ds.load(topo.dp_rank, dp)
    worker = Worker(topo, model, ds, opt)
    sched_cls = SCHEDULES[args.schedule]
    sched = sched_cls(ds.num_mubatches(), pp, topo.stage_id)
    nb = ds.num_batches()

    use_graph = args.graph and not args.no_graph and is_gpu \
        and topo.world == 1

    def one_step(i):
        if use_graph:
            worker.execute_graphed(sched, i % nb)
        else:
            worker.execute(sched, i % nb)

    import torch.distributed as dist

    def barrier_sync():
        if is_gpu:
            torch.cuda.synchronize(device)
        if topo.is_distributed:
            if is_gpu and dist.get_backend() == "nccl":
                dist.barrier(device_ids=[device.index])
            else:
                dist.barrier()
        if is_gpu:
            torch.cuda.synchronize(device)

    for i in range(args.warmup):
        one_step(i)
    barrier_sync()
    t0 = time.perf_counter()
    for i in range(args.steps):
        one_step(i)
    barrier_sync()
    elapsed = time.perf_counter() - t0

    # MAX elapsed over ranks
    if topo.is_distributed:
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=device if is_gpu else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = t.item()

    samples = global_batch * args.steps
    value = samples / elapsed
    ms_per_step = elapsed / args.steps * 1e3

    if topo.rank == 0:
        print(json.dumps({
            "metric": "samples/sec (whole node), 4-layer MLP MNIST-shape, "
                      "DP×PP at 1/2/4/8 GPUs",
            "value": value,
            "unit": "samples/sec",
            "n_gpus": n,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if is_gpu else "fp32",
            "data": "synthetic",
            "config": {
                "model": ("mlp4_mnist_784x256x256x256x10"
                          if sizes == [784, 256, 256, 256, 10] else
                          "mlp%d_%s" % (len(sizes) - 1,
                                        "x".join(map(str, sizes[:3]))
                                        + ("..." if len(sizes) > 4 else ""))),
                "global_batch": global_batch,
                "seq_len": None,
                "parallelism": f"dp{dp}"
                + (f"xpp{pp}" if pp > 1 else "")
                + (f"xtp{tp}" if tp > 1 else ""),
                "schedule": args.schedule,
                "loss": args.loss,
                "mubatches": args.mubatches,
                "hipgraph": use_graph,
                "fp8_fwd": args.fp8_fwd,
                "peak_mem_gb": (round(torch.cuda.max_memory_allocated()
                                      / 2**30, 2) if is_gpu else None),
            },
        }), flush=True)

    if topo.is_distributed:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
