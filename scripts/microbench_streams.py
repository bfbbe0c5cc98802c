"""Quantify multi-stream overlap for the µbatched GEMM sequence.

GPipe/1F1B on one GPU pay a per-kernel latency floor: µ=4 quarters
every fwd/dgrad GEMM (M 16384→4096) but each small kernel still pays
wave ramp + memory latency, so 4 launches cost ~3x one full-size
launch instead of ~4x1/4.  Forward passes of DIFFERENT µbatches are
data-independent, so issuing them on separate HIP streams lets the
ramps overlap.  This microbench measures the headroom before wiring
streams into the Worker: the µ=4 flagship forward-layer sequence
(4 µbatches x 3 hidden layers + first + head shapes) on 1 vs 2 vs 4
streams.

Run (GPU): python scripts/microbench_streams.py
"""

import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from shallowspeed_amd.ops import functional as F  # noqa: E402


def fwd_chain(x, ws, bs):
    h = x
    for w, b in zip(ws, bs):
        h = F.linear_fwd(h, w, b, relu=True)
    return h


def main():
    assert torch.cuda.is_available()
    dev = torch.device("cuda", 0)
    B, MU = 16384, 4
    mb = B // MU
    sizes = [(784, 256), (256, 256), (256, 256), (256, 10)]
    g = torch.Generator(device="cpu").manual_seed(0)
    ws = [torch.randn(o, i, generator=g).bfloat16().to(dev) for i, o in sizes]
    bs = [torch.zeros(o, dtype=torch.bfloat16, device=dev) for _, o in sizes]
    xs = [torch.randn(mb, 784, generator=g).bfloat16().to(dev)
          for _ in range(MU)]
    xfull = torch.randn(B, 784, generator=g).bfloat16().to(dev)

    def run_streams(nstreams, iters=50):
        streams = [torch.cuda.Stream() for _ in range(nstreams)]
        ev_start = torch.cuda.Event(enable_timing=True)
        ev_end = torch.cuda.Event(enable_timing=True)
        # warmup
        for _ in range(5):
            for m in range(MU):
                with torch.cuda.stream(streams[m % nstreams]):
                    fwd_chain(xs[m], ws, bs)
        torch.cuda.synchronize()
        ev_start.record()
        for _ in range(iters):
            for m in range(MU):
                with torch.cuda.stream(streams[m % nstreams]):
                    fwd_chain(xs[m], ws, bs)
        for s in streams:
            torch.cuda.current_stream().wait_stream(s)
        ev_end.record()
        torch.cuda.synchronize()
        return ev_start.elapsed_time(ev_end) / iters * 1e3  # µs

    def run_full(iters=50):
        for _ in range(5):
            fwd_chain(xfull, ws, bs)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(iters):
            fwd_chain(xfull, ws, bs)
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / iters * 1e6

    full = run_full()
    print(f"full-batch fwd (M={B}):        {full:8.1f} µs")
    for ns in (1, 2, 4):
        t = run_streams(ns)
        print(f"µ={MU} fwd on {ns} stream(s):      {t:8.1f} µs "
              f"({t / full:.2f}x full)")


if __name__ == "__main__":
    main()
