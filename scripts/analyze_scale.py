"""Analyze the driver's SCALE_rNN.json (bench at N=1,2,4,8 GPUs).

Computes weak-scaling efficiency per N, estimates the per-step
communication cost from the N=1 baseline, and compares it against the
xGMI ring bound for the flagship's gradient volume — the round-2
RCCL-tuning worksheet (ROADMAP #1).

    python scripts/analyze_scale.py SCALE_r01.json
"""

import json
import sys


def analyze(path):
    runs = json.load(open(path))
    if isinstance(runs, dict):
        runs = runs.get("runs", [runs])
    by_n = {}
    for r in runs:
        n = r.get("n_gpus") or r.get("gpus")
        if n:
            by_n[n] = r
    if 1 not in by_n:
        print("no N=1 baseline in", path)
        return
    base = by_n[1]
    t1 = base["ms_per_step"]
    # flagship gradient volume (4-layer MLP MNIST shape): f32 grads
    grad_bytes = 4 * (784 * 256 + 256 + 256 * 256 + 256 +
                      256 * 256 + 256 + 256 * 10 + 10)
    print(f"N=1: {t1:.4f} ms/step   grad volume {grad_bytes/1e6:.2f} MB")
    for n in sorted(by_n):
        if n == 1:
            continue
        r = by_n[n]
        tn = r["ms_per_step"]
        eff = t1 / tn
        comm = tn - t1  # weak scaling: compute/GPU constant
        # ring all-reduce moves 2(n-1)/n of the volume over the
        # slowest link; xGMI p2p ~153 GB/s/link (guide figure)
        ring_ms = 2 * (n - 1) / n * grad_bytes / 153e9 * 1e3
        print(f"N={n}: {tn:.4f} ms/step  weak-eff {eff:5.1%}  "
              f"apparent comm+sync {comm*1e3:7.1f} µs  "
              f"(xGMI ring bound ≈ {ring_ms*1e3:.0f} µs)")
        if comm > 0.05 and comm > 20 * ring_ms:
            print("      -> latency-bound, not bandwidth-bound: tune "
                  "NCCL_MIN_NCHANNELS / bucket count down, consider "
                  "one-shot allreduce for the small-model regime")


if __name__ == "__main__":
    analyze(sys.argv[1] if len(sys.argv) > 1 else "SCALE_r01.json")
