"""Standalone PyTorch-DDP sanity baseline.

Analog of the reference's scripts/DDP_PyTorch_MNIST.py (167 LoC): a
plain torch MLP trained with data parallelism, used to cross-check the
framework's DP math and to measure distributed-vs-serial weight
divergence (reference :157-167 prints absolute divergence vs a saved
serial checkpoint).

MI355X-native differences: torch.distributed over RCCL ("nccl"
backend) on GPU / gloo on CPU instead of mpi4py; blocking per-param
all-reduce after full backward (the NON-interleaved baseline the
framework's bucketed-overlap reducer is compared against,
reference :119-122); loss rescaled by world size (reference :111-113).

Run:
    python scripts/ddp_pytorch_mnist.py                 # serial, saves ref
    torchrun --standalone --nproc-per-node 2 \
        scripts/ddp_pytorch_mnist.py                    # DDP, prints divergence
"""

import argparse
import os
import sys

import torch
import torch.distributed as dist
import torch.nn as nn

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from shallowspeed_amd.data import synthesize  # noqa: E402


def build_model(seed=7):
    torch.manual_seed(seed)
    return nn.Sequential(
        nn.Linear(784, 256), nn.ReLU(),
        nn.Linear(256, 128), nn.ReLU(),
        nn.Linear(128, 10),
    )


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--epochs", type=int, default=3)
    ap.add_argument("--batch", type=int, default=128)
    ap.add_argument("--lr", type=float, default=0.05)
    ap.add_argument("--samples", type=int, default=4096)
    ap.add_argument("--out", default="data")
    args = ap.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    device = torch.device("cpu")
    if world > 1:
        if torch.cuda.is_available():
            device = torch.device("cuda", int(os.environ.get("LOCAL_RANK", 0)))
            torch.cuda.set_device(device)
            dist.init_process_group("nccl")
        else:
            dist.init_process_group("gloo")

    x, y = synthesize(args.samples, 784, 10, seed=1234)
    labels = y.argmax(1)
    # strided DP shard, like the framework (dataset.py:54-58 semantics)
    xs, ls = x[rank::world].to(device), labels[rank::world].to(device)

    model = build_model().to(device)
    opt = torch.optim.SGD(model.parameters(), lr=args.lr)
    lossf = nn.CrossEntropyLoss()

    nb = xs.shape[0] // (args.batch // world)
    lb = args.batch // world
    for epoch in range(args.epochs):
        for b in range(nb):
            opt.zero_grad()
            xb, lb_ = xs[b * lb:(b + 1) * lb], ls[b * lb:(b + 1) * lb]
            loss = lossf(model(xb), lb_) / world  # reference :111-113
            loss.backward()
            if world > 1:
                # blocking per-param all-reduce after full backward —
                # the non-interleaved DDP baseline (reference :119-122)
                for p in model.parameters():
                    dist.all_reduce(p.grad, op=dist.ReduceOp.SUM)
            opt.step()

    os.makedirs(args.out, exist_ok=True)
    ckpt = os.path.join(args.out, f"ddp_ref_p{world}.pt")
    if rank == 0:
        torch.save(model.state_dict(), ckpt)
        print(f"saved {ckpt}")
    serial = os.path.join(args.out, "ddp_ref_p1.pt")
    if rank == 0 and world > 1 and os.path.exists(serial):
        ref = torch.load(serial, map_location=device, weights_only=True)
        div = max((model.state_dict()[k].float() - ref[k].float())
                  .abs().max().item() for k in ref)
        print(f"max |w_ddp - w_serial| = {div:.3e}")  # reference :159-167
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
