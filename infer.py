"""Batched inference driver (serving path).

Loads a stage-sharded checkpoint (or fresh shape-seeded weights) and
runs the forward-only pipeline over a dataset, reporting predictions
and throughput.  Uses the same Worker + InferenceSchedule machinery as
training eval (reference: compute_accuracy, train.py:21-47, fwd-only
schedule pipe.py:275-294) — so it works single-GPU or as a PP pipeline
under torchrun.

    python infer.py --batch 16384 --samples 65536          # 1 GPU
    torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node 4 \
        infer.py --pp 4 --checkpoint ckpt/                 # pipeline
"""

import argparse
import json
import time

import torch

from shallowspeed_amd.data import Dataset
from shallowspeed_amd.models import MLP
from shallowspeed_amd.parallel import InferenceSchedule, Worker, init_topology


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--pp", type=int, default=1)
    ap.add_argument("--batch", type=int, default=16384)
    ap.add_argument("--samples", type=int, default=65536)
    ap.add_argument("--layer-sizes", default="784,256,256,256,10")
    ap.add_argument("--checkpoint", default=None)
    ap.add_argument("--data-dir", default=None)
    ap.add_argument("--out", default=None, help="write predictions (.pt)")
    ap.add_argument("--device", default=None)
    ap.add_argument("--backend", default=None)
    args = ap.parse_args()

    sizes = [int(s) for s in args.layer_sizes.split(",")]
    while len(sizes) % args.pp != 0:
        sizes.insert(1, sizes[1])

    device = torch.device(args.device) if args.device else None
    topo = init_topology(1, args.pp, backend=args.backend, device=device)
    device = topo.device

    model = MLP(sizes, topo.stage_id, args.pp, args.batch)
    model.materialize_device(device)
    if args.checkpoint:
        from shallowspeed_amd.checkpoint import load_checkpoint

        load_checkpoint(args.checkpoint, model, topo)
    model.eval()

    ds = Dataset(args.batch, args.batch, save_dir=args.data_dir,
                 n_samples=args.samples, in_dim=sizes[0],
                 n_classes=sizes[-1], device=device).load(0, 1)
    worker = Worker(topo, model, ds, None, use_dp=False)
    sched = InferenceSchedule(1, args.pp, topo.stage_id)

    preds = []
    # warmup
    worker.execute(sched, 0)
    if device.type == "cuda":
        torch.cuda.synchronize(device)
    t0 = time.perf_counter()
    for b in range(ds.num_batches()):
        worker.execute(sched, b)
        if topo.stage_id == args.pp - 1:
            # HIP row-argmax, on device; one transfer at the end
            # (torch's ROCm argmax on skinny bf16 is ~1.3 ms/batch)
            from shallowspeed_amd.ops.functional import row_argmax

            preds.append(row_argmax(worker._out_bufs[0]))
    if device.type == "cuda":
        torch.cuda.synchronize(device)
    dt = time.perf_counter() - t0
    preds = [p.cpu() for p in preds]

    if topo.stage_id == args.pp - 1:
        n = ds.num_batches() * args.batch
        print(json.dumps({
            "inference_samples_per_sec": n / dt,
            "ms_per_batch": dt / ds.num_batches() * 1e3,
            "batch": args.batch,
            "n": n,
        }))
        if args.out:
            torch.save(torch.cat(preds), args.out)
            print(f"predictions -> {args.out}")


if __name__ == "__main__":
    main()
