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


def check(M, N, K, bias_on, relu, seed=0, fn=None, tag=""):
    fn = fn or e.gemm_nt_256
    g = torch.Generator(device="cuda").manual_seed(seed)
    a = (torch.rand((M, K), generator=g, device=dev) * 2 - 1).bfloat16()
    b = (torch.rand((N, K), generator=g, device=dev) * 2 - 1).bfloat16()
    bias = ((torch.rand((N,), generator=g, device=dev) * 2 - 1).bfloat16()
            if bias_on else EMPTY)
    c = fn(a, b, bias, relu)
    want = ref(a, b, bias if bias_on else None, relu)
    err = (c.float() - want).abs()
    scale = want.abs().clamp_min(1.0)
    rel = (err / scale).max().item()
    ok = rel < 0.02
    print(f"  {tag}{M}x{N}x{K} bias={bias_on} relu={relu}: "
          f"max_rel={rel:.4f} {'OK' if ok else 'FAIL'}")
    return ok


def race_screen(M, N, K, runs=10, fn=None, tag=""):
    fn = fn or e.gemm_nt_256
    g = torch.Generator(device="cuda").manual_seed(7)
    a = (torch.rand((M, K), generator=g, device=dev) * 2 - 1).bfloat16()
    b = (torch.rand((N, K), generator=g, device=dev) * 2 - 1).bfloat16()
    first = fn(a, b, EMPTY, False).clone()
    ok = True
    for r in range(runs - 1):
        # vary surrounding stream contents to shake scheduling
        _ = torch.randn(512 * (r + 1), 512, device=dev) @ \
            torch.randn(512, 64, device=dev)
        c = fn(a, b, EMPTY, False)
        if not torch.equal(c, first):
            nbad = (c != first).sum().item()
            print(f"  RACE at {tag}{M}x{N}x{K} run {r}: {nbad} mismatches")
            ok = False
    torch.cuda.synchronize()
    print(f"  race screen {tag}{M}x{N}x{K} x{runs}: {'OK' if ok else 'FAIL'}")
    return ok


def v2_suite():
    """refcheck + race + perf for the 4-wave 128x128-wave-tile variant
    (gemm256w.hip) vs the shipped 8-wave kernel and the library."""
    ok = True
    print("== v2 (4-wave) refcheck ==")
    for args in [(256, 256, 128, False, False), (256, 256, 256, True, False),
                 (512, 256, 384, True, True), (256, 512, 512, False, True),
                 (512, 512, 1024, True, False), (768, 256, 2048, True, True),
                 (4096, 4096, 4096, True, False)]:
        ok &= check(*args, fn=e.gemm_nt_256w, tag="v2 ")
    print("== v2 race screen ==")
    ok &= race_screen(512, 512, 1024, fn=e.gemm_nt_256w, tag="v2 ")
    ok &= race_screen(4096, 4096, 4096, runs=6, fn=e.gemm_nt_256w, tag="v2 ")
    print("== v2 perf (v1 | v2 | lib) ==")
    for (M, N, K) in [(4096, 4096, 4096), (8192, 4096, 4096),
                      (16384, 4096, 4096), (16384, 1024, 1024),
                      (16384, 8192, 8192)]:
        a = torch.randn(M, K, device=dev).bfloat16()
        b = torch.randn(N, K, device=dev).bfloat16()
        bt = b.t().contiguous().t()
        fl = 2.0 * M * N * K
        t1 = bench_fn(lambda: e.gemm_nt_256(a, b, EMPTY, False))
        t2 = bench_fn(lambda: e.gemm_nt_256w(a, b, EMPTY, False))
        tl = bench_fn(lambda: a @ bt)
        print(f"  {M}x{N}x{K}: v1 {fl/t1/1e12:7.1f}TF | "
              f"v2 {fl/t2/1e12:7.1f}TF | lib {fl/tl/1e12:7.1f}TF "
              f"(v2/lib {fl/t2/1e12/(fl/tl/1e12)*100:.0f}%)")
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
    ap.add_argument("--v2-only", action="store_true",
                    help="only the 4-wave variant suite")
    args = ap.parse_args()

    ok = True
    if args.v2_only:
        ok = v2_suite()
        print("V2 ALL OK" if ok else "V2 FAILURES", flush=True)
        sys.exit(0 if ok else 1)
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
        print("== fp8 tier ==")
        ok &= f8_quant_roundtrip()
        ok &= f8_check(256, 256, 256)
        ok &= f8_check(512, 256, 384 + 128, seed=2)
        ok &= f8_check(512, 512, 1024, bias_on=True, seed=3)
        ok &= f8_check(768, 512, 2048, relu=True, seed=4)
        ok &= f8_check(4096, 4096, 4096, bias_on=True, seed=5)
        ok &= f8q_check(256, 256, 256)
        ok &= f8q_check(512, 512, 1024, relu=True, seed=2)
        ok &= f8q_check(4096, 4096, 4096, seed=3)
        ok &= f8_race(512, 512, 1024)
        ok &= f8_race(4096, 4096, 4096, runs=4)
        print("== wgrad256 race ==")
        ok &= wg_race(4096, 4096, 4096, bitwise=True)    # split_k == 1
        ok &= wg_race(512, 512, 512, bitwise=False)      # split_k > 1
        ok &= wg_race(16384, 1024, 1024, bitwise=False)
    print("== perf ==")
    perf(4096, 4096, 4096)
    perf(8192, 4096, 4096)
    perf(16384, 4096, 4096)
    perf(16384, 1024, 1024)
    print("== fp8 perf ==")
    f8_perf(4096, 4096, 4096)
    f8_perf(16384, 4096, 4096)
    f8q_perf(16384, 4096, 4096)
    print("== wgrad perf ==")
    wg_perf(16384, 4096, 4096)
    wg_perf(4096, 4096, 4096)
    wg_perf(16384, 1024, 1024)
    print("PASS" if ok else "FAIL")
    sys.exit(0 if ok else 1)




def f8_ref(a, b, bias=None, relu=False):
    c = a.float() @ b.float().t()
    if bias is not None:
        c = c + bias.float()
    if relu:
        c = torch.relu(c)
    return c


def dequant(q, s):
    """Torch-side e4m3+E8M0 dequant — the kernel's exact oracle."""
    R, K = q.shape
    qf = q.to(torch.int32)
    sgn = torch.where(qf >= 128, -1.0, 1.0)
    qa = qf & 0x7F
    ee = qa >> 3
    m = (qa & 7).float()
    mag = torch.where(ee == 0, (m / 8.0) * 2.0 ** -6,
                      (1 + m / 8.0) * torch.pow(2.0, ee.float() - 7))
    v = sgn * mag  # [R][K] dequant WITHOUT block scale
    k = torch.arange(K, device=q.device)
    g = ((k >> 6) & 1) + 2 * ((k >> 4) & 1)          # hw group of each k
    w = k >> 7
    sc = s.to(torch.int32)[w, :, g].t().float()       # [R][K] exponents
    return v * torch.pow(2.0, sc - 127)


def f8_check(M, N, K, bias_on=False, relu=False, seed=0):
    g = torch.Generator(device="cuda").manual_seed(seed)
    a = (torch.rand((M, K), generator=g, device=dev) * 2 - 1).bfloat16()
    b = (torch.rand((N, K), generator=g, device=dev) * 2 - 1).bfloat16()
    bias = ((torch.rand((N,), generator=g, device=dev) * 2 - 1).bfloat16()
            if bias_on else EMPTY)
    qa, sa = e.fp8_quantize(a)
    qb, sb = e.fp8_quantize(b)
    c = e.gemm_nt_f8(qa, sa, qb, sb, bias, relu)
    # exact oracle: f32 matmul of the dequantized operands (isolates
    # kernel correctness from quantization noise)
    want = dequant(qa, sa) @ dequant(qb, sb).t()
    if bias_on:
        want = want + bias.float()
    if relu:
        want = torch.relu(want)
    err = (c.float() - want).abs()
    rel = (err / want.abs().clamp_min(4.0)).max().item()
    # end-to-end (vs unquantized bf16 math): loose sanity
    full = f8_ref(a, b, bias if bias_on else None, relu)
    e2e = ((c.float() - full).abs() /
           full.abs().clamp_min(float(full.abs().mean()))).max().item()
    ok = rel < 0.02
    print(f"  fp8 {M}x{N}x{K} bias={bias_on} relu={relu}: kernel_rel="
          f"{rel:.5f} e2e_quant_rel={e2e:.3f} {'OK' if ok else 'FAIL'}")
    return ok


def f8_quant_roundtrip():
    g = torch.Generator(device="cuda").manual_seed(1)
    x = (torch.randn((64, 256), generator=g, device=dev) * 3).bfloat16()
    q, s = e.fp8_quantize(x)
    # dequantize on host and compare (per-element fp8 quantization err)
    qf = q.cpu().view(torch.uint8).float()
    ok = True
    import struct

    def e4m3(b):
        sgn = -1.0 if b & 0x80 else 1.0
        ee = (b >> 3) & 0xF
        m = b & 7
        if ee == 0:
            return sgn * (m / 8.0) * 2.0 ** -6
        return sgn * (1 + m / 8.0) * 2.0 ** (ee - 7)

    xe = x.float().cpu()
    se = s.cpu()
    err = 0.0
    for r in range(0, 64, 17):
        for k in range(0, 256, 37):
            gg = ((k % 128) >> 6 & 1) + 2 * ((k >> 4) & 1)
            sc = 2.0 ** (int(se[k // 128, r, gg]) - 127)
            v = e4m3(int(q[r, k])) * sc
            err = max(err, abs(v - float(xe[r, k])) / max(abs(float(xe[r, k])), 1.0))
    ok = err < 0.08
    print(f"  fp8 quant roundtrip: max_rel={err:.4f} {'OK' if ok else 'FAIL'}")
    return ok


def f8q_check(M, N, K, relu=False, seed=0):
    """Fused-quant output: dequant(q,s) must match the bf16-out path
    within one e4m3 quantization step."""
    g = torch.Generator(device="cuda").manual_seed(seed)
    a = (torch.rand((M, K), generator=g, device=dev) * 2 - 1).bfloat16()
    b = (torch.rand((N, K), generator=g, device=dev) * 2 - 1).bfloat16()
    qa, sa = e.fp8_quantize(a)
    qb, sb = e.fp8_quantize(b)
    c16 = e.gemm_nt_f8(qa, sa, qb, sb, EMPTY, relu).float()
    cq, cs = e.gemm_nt_f8_q(qa, sa, qb, sb, EMPTY, relu)
    cd = dequant(cq, cs)
    rel = ((cd - c16).abs() / c16.abs().clamp_min(4.0)).max().item()
    ok = rel < 0.07  # one e4m3 ulp
    print(f"  f8q fused-out {M}x{N}x{K} relu={relu}: rel={rel:.4f} "
          f"{'OK' if ok else 'FAIL'}")
    return ok


def f8q_perf(M, N, K):
    a = torch.randn(M, K, device=dev).bfloat16()
    b = torch.randn(N, K, device=dev).bfloat16()
    qa, sa = e.fp8_quantize(a)
    qb, sb = e.fp8_quantize(b)
    fl = 2.0 * M * N * K
    tq = bench_fn(lambda: e.gemm_nt_f8_q(qa, sa, qb, sb, EMPTY, True))
    t16 = bench_fn(lambda: e.gemm_nt_f8(qa, sa, qb, sb, EMPTY, True))
    print(f"  f8q {M}x{N}x{K}: fused-q-out {tq*1e6:7.1f}us "
          f"{fl/tq/1e12:7.1f}TF | bf16-out {t16*1e6:7.1f}us "
          f"{fl/t16/1e12:7.1f}TF")


def f8_race(M, N, K, runs=8):
    g = torch.Generator(device="cuda").manual_seed(3)
    a = (torch.rand((M, K), generator=g, device=dev) * 2 - 1).bfloat16()
    b = (torch.rand((N, K), generator=g, device=dev) * 2 - 1).bfloat16()
    qa, sa = e.fp8_quantize(a)
    qb, sb = e.fp8_quantize(b)
    first = e.gemm_nt_f8(qa, sa, qb, sb, EMPTY, False).clone()
    ok = True
    for r in range(runs - 1):
        _ = torch.randn(512, 512, device=dev) @ torch.randn(512, 64, device=dev)
        if not torch.equal(e.gemm_nt_f8(qa, sa, qb, sb, EMPTY, False), first):
            ok = False
    torch.cuda.synchronize()
    print(f"  fp8 race {M}x{N}x{K} x{runs}: {'OK' if ok else 'FAIL'}")
    return ok


def f8_perf(M, N, K):
    a = torch.randn(M, K, device=dev).bfloat16()
    b = torch.randn(N, K, device=dev).bfloat16()
    qa, sa = e.fp8_quantize(a)
    qb, sb = e.fp8_quantize(b)
    fl = 2.0 * M * N * K
    t8 = bench_fn(lambda: e.gemm_nt_f8(qa, sa, qb, sb, EMPTY, False))
    tq = bench_fn(lambda: e.fp8_quantize(a))
    t16 = bench_fn(lambda: e.gemm_nt_256(a, b, EMPTY, False))
    print(f"  fp8 {M}x{N}x{K}: {t8*1e6:7.1f}us {fl/t8/1e12:7.1f}TF | "
          f"quantA {tq*1e6:6.1f}us | bf16-8phase {t16*1e6:7.1f}us "
          f"{fl/t16/1e12:7.1f}TF")


if __name__ == "__main__":
    main()
