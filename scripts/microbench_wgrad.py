"""Microbench wgrad shapes/splits on MI355X (run under gpurun)."""
import sys
import time

import torch

sys.path.insert(0, ".")
from shallowspeed_amd.ops import load_ext  # noqa: E402

e = load_ext(required=True)
dev = torch.device("cuda", 0)
empty = torch.Tensor()


def bench(fn, iters=200):
    for _ in range(20):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6  # us


shapes = [(16384, 256, 784), (16384, 256, 256), (16384, 10, 256)]
for Kb, Mo, N in shapes:
    dy = torch.randn(Kb, Mo, device=dev).bfloat16()
    x = torch.randn(Kb, N, device=dev).bfloat16()
    mask = torch.randn(Kb, Mo, device=dev).bfloat16()
    gw = torch.zeros(Mo, N, device=dev, dtype=torch.float32)
    gb = torch.zeros(Mo, device=dev, dtype=torch.float32)
    mb = (Kb * (Mo + N) * 2 + (Kb * Mo * 2)) / 1e6
    for sk in (0, 8, 16, 32, 64, 128):
        us = bench(lambda: e.wgrad_tn(dy, x, gw, gb, mask, sk))
        print(f"wgrad Kb={Kb} Mo={Mo} N={N} split={sk:3d}: {us:7.1f} us "
              f"({mb/us*1e3:.0f} GB/s algorithmic)")
    us = bench(lambda: e.wgrad_tn(dy, x, gw, empty, mask, 0))
    print(f"  no-db: {us:7.1f} us")
    us = bench(lambda: e.wgrad_tn(dy, x, gw, empty, empty, 0))
    print(f"  no-db no-mask: {us:7.1f} us")

# reference comparison: torch matmul does the same math (library GEMM)
dy = torch.randn(16384, 256, device=dev).bfloat16()
x = torch.randn(16384, 784, device=dev).bfloat16()
us = bench(lambda: dy.t() @ x)
print(f"torch bf16 dy.t()@x (16384,256,784): {us:.1f} us")
us = bench(lambda: torch.matmul(dy.t(), x))
print(f"torch matmul again: {us:.1f} us")

# forward gemm for comparison
w = torch.randn(256, 784, device=dev).bfloat16()
b = torch.randn(256, device=dev).bfloat16()
xx = torch.randn(16384, 784, device=dev).bfloat16()
us = bench(lambda: e.gemm_nt(xx, w, b, empty, True))
print(f"gemm_nt fwd (16384,256,784)+bias+relu: {us:.1f} us")
us = bench(lambda: xx @ w.t())
print(f"torch xx@w.t(): {us:.1f} us")
