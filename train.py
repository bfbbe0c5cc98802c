"""CLI training driver.

Reference: /root/reference/train.py — identical core CLI
(`--dp N --pp M --schedule {naive,gpipe,pipedream}`, train.py:62-74)
plus flags for batch/µbatch/layer/loss/dtype/synthetic-data knobs
(the reference hardwires EPOCHS=20, GLOBAL_BATCH_SIZE=128,
N_MUBATCHES=4, lr=0.006, layer_sizes=[784,128,...,10] at
train.py:56-59,98,107).

Launch (multi-GPU, one rank per GPU over RCCL):
    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
        --master-addr 127.0.0.1 train.py --dp D --pp P ...   (D*P == N)

Single process: python train.py          (dp=1, pp=1, CPU or 1 GPU)
"""

import argparse
import time

import torch

from shallowspeed_amd.data import Dataset
from shallowspeed_amd.models import MLP, SGD
from shallowspeed_amd.parallel import SCHEDULES, InferenceSchedule, Worker, init_topology
from shallowspeed_amd.utils import assert_sync, get_model_hash, rprint


def compute_accuracy(model, worker, dataset, topo):
    """Forward-only eval over the val split; returns accuracy on the
    last stage, None elsewhere.  Reference: train.py:21-47 (argmax of
    the output buffer vs argmax of target, train.py:40-43)."""
    model.eval()
    correct, total = 0, 0
    sched = InferenceSchedule(1, topo.pp, topo.stage_id)
    for b in range(dataset.num_batches()):
        worker.execute(sched, b)
        if topo.stage_id == topo.pp - 1:
            from shallowspeed_amd.ops.functional import row_argmax

            probs = worker._out_bufs[0]
            target = dataset.micro_batch_target(b, 0)
            pred = row_argmax(probs)
            lab = row_argmax(target.to(probs.device))
            correct += (pred == lab).sum().item()
            total += pred.numel()
    model.train()
    if topo.stage_id != topo.pp - 1:
        return None
    if topo.dp > 1:
        t = torch.tensor([correct, total], dtype=torch.float64)
        import torch.distributed as dist

        if topo.device.type == "cuda":
            t = t.to(topo.device)
        dist.all_reduce(t, group=topo.dp_group)
        correct, total = t[0].item(), t[1].item()
    return correct / max(total, 1)


def parse_sizes(s):
    return [int(v) for v in s.split(",")]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--dp", type=int, default=1)
    ap.add_argument("--pp", type=int, default=1)
    ap.add_argument("--tp", type=int, default=1,
                    help="tensor parallelism (Megatron-style column/row "
                         "pair, beyond-reference); composes with --dp "
                         "(TP innermost = xGMI-adjacent ranks)")
    ap.add_argument("--schedule", choices=list(SCHEDULES), default="gpipe")
    ap.add_argument("--epochs", type=int, default=20)
    ap.add_argument("--global-batch", type=int, default=128)
    ap.add_argument("--mubatches", type=int, default=4)
    ap.add_argument("--lr", type=float, default=0.006)
    ap.add_argument("--momentum", type=float, default=0.0)
    ap.add_argument("--weight-decay", type=float, default=0.0)
    ap.add_argument("--optimizer", choices=["sgd", "adamw"], default="sgd",
                    help="sgd matches the reference (optimizer.py:4-13); "
                         "adamw is the fused multi-tensor AdamW extension")
    ap.add_argument("--clip-grad", type=float, default=None,
                    help="global-norm gradient clipping threshold "
                         "(applied on the flat grad buffer after the DP "
                         "all-reduce, before the optimizer step)")
    ap.add_argument("--lr-schedule", choices=["const", "cosine"],
                    default="const",
                    help="per-epoch LR schedule: const (reference "
                         "behavior) or cosine decay to 10%% of --lr")
    ap.add_argument("--warmup-epochs", type=int, default=0,
                    help="linear LR warmup epochs before the schedule")
    ap.add_argument("--deterministic", action="store_true",
                    help="bitwise run-to-run reproducible GPU training "
                         "(forces the single-owner split_k=1 wgrad path; "
                         "bucket reduction order is fixed by construction)")
    ap.add_argument("--fp8-fwd", action="store_true",
                    help="MX-fp8 training forward on qualifying Linears "
                         "(O%%256,I%%256,µbatch%%256); backward stays bf16, "
                         "masters f32")
    ap.add_argument("--layer-sizes", type=parse_sizes,
                    default=[784, 128, 127, 126, 125, 124, 123, 10],
                    help="comma-separated boundaries (reference train.py:98)")
    ap.add_argument("--loss", choices=["xent", "mse"], default="mse",
                    help="loss head; mse matches the reference head")
    ap.add_argument("--data-dir", default=None,
                    help="load x_*.parquet / y_*.npy like the reference "
                         "if present; otherwise synthetic")
    ap.add_argument("--samples", type=int, default=8192)
    ap.add_argument("--val-batch", type=int, default=None,
                    help="validation batch size (default: auto — whole "
                         "val split when --data-dir files exist)")
    ap.add_argument("--backend", default=None,
                    help="torch.distributed backend override "
                         "(default: nccl on GPU, gloo on CPU)")
    ap.add_argument("--device", default=None, help="cpu | cuda")
    ap.add_argument("--save", default=None, help="checkpoint dir to write")
    ap.add_argument("--timing", action="store_true",
                    help="print per-instruction wall time after training")
    ap.add_argument("--resume", default=None, help="checkpoint dir to read")
    args = ap.parse_args()

    assert args.global_batch % (args.dp * args.mubatches) == 0, \
        "--global-batch must divide by dp*mubatches"
    assert len(args.layer_sizes) % args.pp == 0, \
        "len(--layer-sizes) must divide into --pp stages (layers.py:242)"
    device = torch.device(args.device) if args.device else None
    topo = init_topology(args.dp, args.pp, backend=args.backend,
                         device=device, tp=args.tp)
    device = topo.device

    # model: stage slice of the full MLP (reference train.py:99-107)
    if args.deterministic:
        from shallowspeed_amd.ops import functional as _F

        _F.set_deterministic(True)

    if args.tp > 1:
        from shallowspeed_amd.parallel import TPMLP

        model = TPMLP(args.layer_sizes, topo.tp_group, topo.tp_rank,
                      args.tp, args.global_batch, loss=args.loss)
    else:
        model = MLP(args.layer_sizes, stage_idx=topo.stage_id,
                    n_stages=args.pp,
                    global_batch_size=args.global_batch, loss=args.loss)
    model.materialize_device(device)
    clip_kw = dict(clip_norm=args.clip_grad, flat_grad=model._flat_grad)
    if args.optimizer == "adamw":
        from shallowspeed_amd.models import AdamW

        optimizer = AdamW(model.parameters(), lr=args.lr,
                          weight_decay=args.weight_decay, **clip_kw)
    else:
        optimizer = SGD(model.parameters(), lr=args.lr,
                        momentum=args.momentum,
                        weight_decay=args.weight_decay, **clip_kw)

    if args.resume:
        from shallowspeed_amd.checkpoint import load_checkpoint

        load_checkpoint(args.resume, model, topo, optimizer=optimizer)

    if args.fp8_fwd:
        n = model.set_fp8_fwd(True)
        rprint(f"MX-fp8 training forward enabled on {n} layer(s)")

    mubatch = args.global_batch // args.dp // args.mubatches
    train_ds = Dataset(args.global_batch, mubatch, save_dir=args.data_dir,
                       n_samples=args.samples, in_dim=args.layer_sizes[0],
                       n_classes=args.layer_sizes[-1], device=device)
    train_ds.load(topo.dp_rank, args.dp)
    val_batch = args.val_batch
    if val_batch is None and args.data_dir:
        import os

        yp = os.path.join(args.data_dir, "y_val.npy")
        if os.path.exists(yp):
            import numpy as np

            n_val = np.load(yp, mmap_mode="r").shape[0]
            val_batch = max(1, min(n_val // args.dp, 2048))
    if val_batch is None:
        val_batch = min(args.samples // 4, 1024)
    val_ds = Dataset(val_batch * args.dp, val_batch, save_dir=args.data_dir,
                     validation=True, n_samples=max(args.samples // 4, val_batch * args.dp),
                     in_dim=args.layer_sizes[0],
                     n_classes=args.layer_sizes[-1], device=device)
    val_ds.load(topo.dp_rank, args.dp)

    worker = Worker(topo, model, train_ds, optimizer)
    if args.timing:
        worker.enable_instruction_timing(True)
    val_worker = Worker(topo, model, val_ds, None, use_dp=False)
    sched_cls = SCHEDULES[args.schedule]

    rprint(f"world={topo.world} dp={args.dp} pp={args.pp} "
           f"schedule={args.schedule} device={device} "
           f"layers={args.layer_sizes} loss={args.loss}")

    def lr_at(epoch):
        """Linear warmup then const/cosine decay (to 10% of peak)."""
        if args.warmup_epochs and epoch < args.warmup_epochs:
            return args.lr * (epoch + 1) / args.warmup_epochs
        if args.lr_schedule == "cosine":
            import math

            span = max(1, args.epochs - args.warmup_epochs)
            t = (epoch - args.warmup_epochs) / span
            return args.lr * (0.1 + 0.9 * 0.5 * (1 + math.cos(math.pi * t)))
        return args.lr

    for epoch in range(args.epochs):
        t0 = time.time()
        optimizer.lr = lr_at(epoch)
        acc = compute_accuracy(model, val_worker, val_ds, topo)
        for batch_id in range(train_ds.num_batches()):
            sched = sched_cls(train_ds.num_mubatches(), args.pp, topo.stage_id)
            worker.execute(sched, batch_id)
        if device.type == "cuda":
            torch.cuda.synchronize(device)
        if topo.stage_id == topo.pp - 1 and topo.dp_rank == 0 \
                and (topo.tp == 1 or topo.tp_rank == 0):
            print(f"epoch {epoch:3d}  val_acc={acc:.4f}  "
                  f"time={time.time()-t0:.2f}s", flush=True)

    acc = compute_accuracy(model, val_worker, val_ds, topo)
    if topo.stage_id == topo.pp - 1 and topo.dp_rank == 0 \
            and (topo.tp == 1 or topo.tp_rank == 0):
        print(f"final val_acc={acc:.4f}", flush=True)

    if args.save:
        from shallowspeed_amd.checkpoint import save_checkpoint

        save_checkpoint(args.save, model, topo, step=args.epochs,
                        optimizer=optimizer)

    if args.timing and topo.rank == 0:
        total = sum(worker.instruction_times.values())
        print("instruction time breakdown:")
        for k, v in sorted(worker.instruction_times.items(),
                           key=lambda kv: -kv[1]):
            print(f"  {k:24s} {v:8.3f}s ({v/total*100:5.1f}%)")

    # replica-sync invariant (reference train.py:154-155).  TP ranks
    # hold different shards by design — no hash sync there.
    if topo.dp > 1:
        assert_sync(topo.dp_group, get_model_hash(model))
        rprint("DP replicas in sync ✓")


if __name__ == "__main__":
    main()
