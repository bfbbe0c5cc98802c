"""GPU validation + A/B for the 256-tile 8-phase GEMM (gemm256.hip).

New-template discipline (cdna guide): refcheck at small/odd shapes,
multi-run race screen, then within-probe perf A/B vs the shipped glds
tier and the library GEMM.  Run on a box:

    python scripts/test_gemm256.py [--perf-only]
"""

import argparse
import sys
import time

import torch

sys.path.insert(0, ".")
from shallowspeed_amd.ops import load_ext  # noqa: E402

e = load_ext(required=True)
dev = torch.device("cuda", 0)
EMPTY = torch.Tensor()


def ref(a, b, bias=None, relu=False):
    c = a.float() @ b.float().t()
    if bias is not None:
        c += bias.float()
    if relu:
        c = torch.relu(c)
    return c


def check(M, N, K, bias_on, relu, seed=0):
    g = torch.Generator(device="cuda").manual_seed(seed)
    a = (torch.rand((M, K), generator=g, device=dev) * 2 - 1).bfloat16()
    b = (torch.rand((N, K), generator=g, device=dev) * 2 - 1).bfloat16()
    bias = ((torch.rand((N,), generator=g, device=dev) * 2 - 1).bfloat16()
            if bias_on else EMPTY)
    c = e.gemm_nt_256(a, b, bias, relu)
    want = ref(a, b, bias if bias_on else None, relu)
    err = (c.float() - want).abs()
    scale = want.abs().clamp_min(1.0)
    rel = (err / scale).max().item()
    ok = rel < 0.02
    print(f"  {M}x{N}x{K} bias={bias_on} relu={relu}: "
          f"max_rel={rel:.4f} {'OK' if ok else 'FAIL'}")
    return ok


def race_screen(M, N, K, runs=10):
    g = torch.Generator(device="cuda").manual_seed(7)
    a = (torch.rand((M, K), generator=g, device=dev) * 2 - 1).bfloat16()
    b = (torch.rand((N, K), generator=g, device=dev) * 2 - 1).bfloat16()
    first = e.gemm_nt_256(a, b, EMPTY, False).clone()
    ok = True
    for r in range(runs - 1):
        # vary surrounding stream contents to shake scheduling
        _ = torch.randn(512 * (r + 1), 512, device=dev) @ \
            torch.randn(512, 64, device=dev)
        c = e.gemm_nt_256(a, b, EMPTY, False)
        if not torch.equal(c, first):
            nbad = (c != first).sum().item()
            print(f"  RACE at {M}x{N}x{K} run {r}: {nbad} mismatches")
            ok = False
    torch.cuda.synchronize()
    print(f"  race screen {M}x{N}x{K} x{runs}: {'OK' if ok else 'FAIL'}")
    return ok


def bench_fn(fn, iters=30, warm=5):
    for _ in range(warm):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def perf(M, N, K):
    a = torch.randn(M, K, device=dev).bfloat16()
    b = torch.randn(N, K, device=dev).bfloat16()
    bt = b.t().contiguous().t()  # for matmul column layout fairness
    fl = 2.0 * M * N * K
    t256 = bench_fn(lambda: e.gemm_nt_256(a, b, EMPTY, False))
    told = bench_fn(lambda: e.gemm_nt(a, b, EMPTY, EMPTY, False))
    tlib = bench_fn(lambda: a @ bt)
    print(f"  {M}x{N}x{K}: 8phase {t256*1e6:8.1f}us {fl/t256/1e12:7.1f}TF | "
          f"glds128 {told*1e6:8.1f}us {fl/told/1e12:7.1f}TF | "
          f"lib {tlib*1e6:8.1f}us {fl/tlib/1e12:7.1f}TF")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--perf-only", action="store_true")
    args = ap.parse_args()

    ok = True
    if not args.perf_only:
        print("== refcheck ==")
        ok &= check(256, 256, 128, False, False)
        ok &= check(256, 256, 256, True, False)
        ok &= check(512, 256, 384, True, True)
        ok &= check(256, 512, 512, False, True)
        ok &= check(512, 512, 1024, True, False)
        ok &= check(768, 256, 2048, True, True, seed=3)
        ok &= check(4096, 4096, 4096, True, False, seed=5)
        print("== race screen ==")
        ok &= race_screen(256, 256, 512)
        ok &= race_screen(512, 512, 1024)
        ok &= race_screen(4096, 4096, 4096, runs=6)
    print("== perf ==")
    perf(4096, 4096, 4096)
    perf(8192, 4096, 4096)
    perf(16384, 4096, 4096)
    perf(16384, 1024, 1024)
    print("PASS" if ok else "FAIL")
    sys.exit(0 if ok else 1)


if __name__ == "__main__":
    main()
