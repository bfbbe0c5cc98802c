"""GPU validation + A/B for the 256-tile 8-phase GEMM (gemm256.hip).

New-template discipline (cdna guide): refcheck at small/odd shapes,
multi-run race screen, then within-probe perf A/B vs the shipped glds
tier and the library GEMM.  Run on a box:

    python scripts/test_gemm256.py [--perf-only]
"""

import argparse
import os
import sys
import time

import torch

# compare against the genuine old tiers in the perf columns
os.environ["SS_GEMM256"] = "0"
os.environ["SS_WGRAD256"] = "0"

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


def wg_check(Kb, Mo, N, seed=0):
    g = torch.Generator(device="cuda").manual_seed(seed)
    dy = (torch.rand((Kb, Mo), generator=g, device=dev) * 2 - 1).bfloat16()
    x = (torch.rand((Kb, N), generator=g, device=dev) * 2 - 1).bfloat16()
    gw = torch.randn(Mo, N, generator=g, device=dev)  # pre-seeded: += check
    want = gw + dy.float().t() @ x.float()
    e.wgrad_tn_256(dy, x, gw)
    err = ((gw - want).abs() / want.abs().clamp_min(8.0)).max().item()
    ok = err < 0.02
    print(f"  wgrad {Kb}x{Mo}x{N}: max_rel={err:.4f} {'OK' if ok else 'FAIL'}")
    return ok


def wg_race(Kb, Mo, N, runs=8, bitwise=True):
    """bitwise=True only for split_k==1 shapes; split-K f32 atomics
    accumulate in nondeterministic order (expected, like the 128-tile
    split-K kernel) so those compare within tolerance."""
    g = torch.Generator(device="cuda").manual_seed(9)
    dy = (torch.rand((Kb, Mo), generator=g, device=dev) * 2 - 1).bfloat16()
    x = (torch.rand((Kb, N), generator=g, device=dev) * 2 - 1).bfloat16()
    outs = []
    for r in range(runs):
        gw = torch.zeros(Mo, N, device=dev)
        _ = torch.randn(256 * (r + 1), 256, device=dev).sum()
        e.wgrad_tn_256(dy, x, gw)
        outs.append(gw)
    if bitwise:
        ok = all(torch.equal(o, outs[0]) for o in outs[1:])
    else:
        ok = all(torch.allclose(o, outs[0], rtol=1e-5, atol=1e-3)
                 for o in outs[1:])
    torch.cuda.synchronize()
    print(f"  wgrad race {Kb}x{Mo}x{N} x{runs} "
          f"({'bitwise' if bitwise else 'tol'}): {'OK' if ok else 'FAIL'}")
    return ok


def wg_perf(Kb, Mo, N):
    dy = torch.randn(Kb, Mo, device=dev).bfloat16()
    x = torch.randn(Kb, N, device=dev).bfloat16()
    gw = torch.zeros(Mo, N, device=dev)
    gb = torch.zeros(Mo, device=dev)
    fl = 2.0 * Kb * Mo * N
    t256 = bench_fn(lambda: e.wgrad_tn_256(dy, x, gw))
    told = bench_fn(lambda: e.wgrad_tn(dy, x, gw, torch.Tensor(), torch.Tensor(), 0))
    tlib = bench_fn(lambda: dy.t().float() @ x.float() if False else dy.t() @ x)
    print(f"  wgrad {Kb}x{Mo}x{N}: 8phase {t256*1e6:8.1f}us {fl/t256/1e12:7.1f}TF | "
          f"wgrad128 {told*1e6:8.1f}us {fl/told/1e12:7.1f}TF | "
          f"lib(bf16 out) {tlib*1e6:8.1f}us {fl/tlib/1e12:7.1f}TF")


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
        print("== wgrad256 refcheck ==")
        ok &= wg_check(128, 256, 256)
        ok &= wg_check(384, 512, 256, seed=2)
        ok &= wg_check(1024, 256, 768, seed=3)
        ok &= wg_check(2048, 1024, 1024, seed=4)
        ok &= wg_check(16384, 1024, 1024, seed=5)
        ok &= wg_check(4096, 4096, 4096, seed=6)
        print("== wgrad256 race ==")
        ok &= wg_race(4096, 4096, 4096, bitwise=True)    # split_k == 1
        ok &= wg_race(512, 512, 512, bitwise=False)      # split_k > 1
        ok &= wg_race(16384, 1024, 1024, bitwise=False)
    print("== perf ==")
    perf(4096, 4096, 4096)
    perf(8192, 4096, 4096)
    perf(16384, 4096, 4096)
    perf(16384, 1024, 1024)
    print("== wgrad perf ==")
    wg_perf(16384, 4096, 4096)
    wg_perf(4096, 4096, 4096)
    wg_perf(16384, 1024, 1024)
    print("PASS" if ok else "FAIL")
    sys.exit(0 if ok else 1)


if __name__ == "__main__":
    main()
