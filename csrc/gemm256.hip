// 256×256-tile 8-phase MFMA bf16 GEMM — the compute-bound tier for
// wide layers (gfx950 / MI355X).
//
//   C[M,N] = A[M,K] · B[N,K]^T (+bias) (+ReLU)   M%256==0, N%256==0,
//                                                K%128==0, K>=128.
//
// This is the deep-pipeline structure the 128²/2-barrier glds kernel
// (gemm.hip) cannot reach: 8 waves (2M×4N, wave tile 128×64,
// acc[8][4] = 128 VGPRs), BK=64, 2 K-tiles per unrolled iteration,
// 8 phases of {2×glds | pre-barrier ds_read prefetch | counted
// lgkmcnt | 16 MFMA under setprio(1) | barrier}, with an extra
// vmcnt(4)+barrier certification point at P4/P8 only (barriers
// elsewhere are thinned: the phase-end barrier alone carries the
// freeing proof).  LDS = 128 KiB: 2 buffers ×
// (A 256×64 + B 256×64) bf16 — legal as a single static allocation on
// gfx950 (160 KiB/CU).
//
// Correctness is by a statically-derived schedule (no dynamic sync):
//   * staging granule = 64 rows × 128 B; each wave stages an 8-row
//     slice per glds (64 lanes × 16 B, lane-linear LDS).
//   * uniform 2 glds per wave per phase (P1,P2: this iteration's
//     second tile's B | P3,P4: next tile's A | P5,P6: its B | P7,P8:
//     the tile after's A);
//     `s_waitcnt vmcnt(4)` at P4/P8 certifies every granule ≥2-3
//     phases before its first ds_read (ledger in the loop comments).
//   * fragment ds_reads run ONE phase ahead of their MFMA consumer
//     with counted lgkmcnt (never 0 in the loop), so a K-tile's LDS
//     is fully read two phases into its 4-phase window — freeing its
//     granules for overwrite exactly when the staging schedule
//     arrives.
//   * LDS slot swizzle: 16-B slot s of row r lives at physical slot
//     s ^ ((r>>1)&7) — a 3-bit bijection that makes BOTH the
//     ds_read_b128 fragment pattern and the lane-linear glds landing
//     conflict-free (the fragment read's 16-lane groups then touch
//     all 64 banks exactly once; rule-21: swizzle applied to the glds
//     SOURCE address and the read address, never the LDS dest).
//
// Reference scope: this serves the same linear-forward/dgrad ops as
// functional.py:13-21 of the reference, at wide-model shapes.

#include "common.h"

namespace {

typedef __attribute__((address_space(1))) const unsigned int* gptr_t;
typedef __attribute__((address_space(3))) unsigned int* lptr_t;

#define SS_VMCNT(n) asm volatile("s_waitcnt vmcnt(%0)" ::"n"(n))
#define SS_LGKM(n) asm volatile("s_waitcnt lgkmcnt(%0)" ::"n"(n))
#define SS_BAR() __builtin_amdgcn_s_barrier()

template <bool HAS_BIAS, bool RELU>
__global__ __launch_bounds__(512, 1) void gemm_nt_256_kernel(
    const __bf16* __restrict__ A,     // [M][K]
    const __bf16* __restrict__ B,     // [N][K]
    const __bf16* __restrict__ bias,  // [N]
    __bf16* __restrict__ C,           // [M][N]
    int M, int N, int K) {
    // [op A=0/B=1][buf][256 rows × 64 k] bf16 = 128 KiB.  Op-major so
    // every ds_read is base-VGPR + a <64 KiB immediate (buf stride
    // 32 KiB): the whole loop runs on 6 per-lane address registers.
    __shared__ ushort lds[2][2][256 * 64];

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wave = tid >> 6;  // 0..7
    const int wm = wave >> 2;   // 2 M-halves
    const int wn = wave & 3;    // 4 N-quarters

    // XCD-aware bijective block remap (8 XCDs, per-XCD L2)
    const int gx = gridDim.x, gy = gridDim.y;
    const int nwg = gx * gy;
    const int hw = blockIdx.x + gx * blockIdx.y;
    const int xcd = hw % 8, q8 = nwg / 8, r8 = nwg % 8;
    const int wid = (xcd < r8 ? xcd * (q8 + 1)
                              : r8 * (q8 + 1) + (xcd - r8) * q8) + hw / 8;
    const int m0 = ((wid / gy) % gx) * 256;
    const int n0 = (wid % gy) * 256;

    // ---- glds staging constants ----------------------------------
    // Wave w stages rows [g*64 + w*8, +8) of granule g; lane covers
    // (row = lane>>3 in the slice, physical slot = lane&7).  Source
    // slot is pre-swizzled: s = sig ^ ((row_in_tile>>1)&7); only the
    // wave/lane bits survive the &7 (g*64 ≡ 0 mod 16 rows).
    const int lrow8 = lane >> 3;
    const int sig = lane & 7;
    const int slog = sig ^ (((wave & 1) << 2) | (lrow8 >> 1));
    // 32-bit per-lane byte offsets against the uniform A/B pointers
    // (saddr+voffset form; M·K·2 ≤ 2³¹ is enforced by the launcher).
    const int avoff = (m0 + wave * 8 + lrow8) * K * 2 + slog * 16;
    const int bvoff = (n0 + wave * 8 + lrow8) * K * 2 + slog * 16;
    const int gstride = 64 * K * 2;  // granule row stride (bytes)

    auto stageA = [&](int buf, int g, int t) {
        __builtin_amdgcn_global_load_lds(
            (gptr_t)((const char*)A + (avoff + g * gstride + t * 128)),
            (lptr_t)((char*)&lds[0][buf][0] + (g * 64 + wave * 8) * 128),
            16, 0, 0);
    };
    auto stageB = [&](int buf, int g, int t) {
        __builtin_amdgcn_global_load_lds(
            (gptr_t)((const char*)B + (bvoff + g * gstride + t * 128)),
            (lptr_t)((char*)&lds[1][buf][0] + (g * 64 + wave * 8) * 128),
            16, 0, 0);
    };

    // ---- fragment reads (ds_read_b128, swizzled) ------------------
    const int lrow = lane & 15;
    const int kch = lane >> 4;
    const int phys0 = (kch ^ (lrow >> 1)) * 16;  // kh=0 slot byte off
    // 4 per-lane LDS base pointers; every read below is base + a
    // compile-time immediate (buf·32 KiB + frag offset ≤ 47 KiB).
    const char* rdA0 = (const char*)&lds[0][0][0] + (wm * 128 + lrow) * 128 + phys0;
    const char* rdA1 = (const char*)&lds[0][0][0] + (wm * 128 + lrow) * 128 + (phys0 ^ 64);
    const char* rdB0 = (const char*)&lds[1][0][0] + (wn * 64 + lrow) * 128 + phys0;
    const char* rdB1 = (const char*)&lds[1][0][0] + (wn * 64 + lrow) * 128 + (phys0 ^ 64);
    bf16x8 Af[2][4][2];  // [msub][i][kh]
    bf16x8 Bf[2][2][2];  // [nsub][j][kh]

    auto readA = [&](int buf, int msub) {
#pragma unroll
        for (int i = 0; i < 4; ++i) {
            const int off = buf * 32768 + (msub * 64 + i * 16) * 128;
            Af[msub][i][0] = *(const bf16x8*)(rdA0 + off);
            Af[msub][i][1] = *(const bf16x8*)(rdA1 + off);
        }
    };
    auto readB = [&](int buf, int nsub) {
#pragma unroll
        for (int j = 0; j < 2; ++j) {
            const int off = buf * 32768 + (nsub * 32 + j * 16) * 128;
            Bf[nsub][j][0] = *(const bf16x8*)(rdB0 + off);
            Bf[nsub][j][1] = *(const bf16x8*)(rdB1 + off);
        }
    };

    f32x4 acc[8][4];
#pragma unroll
    for (int i = 0; i < 8; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

    auto mfma16 = [&](int msub, int nsub) {
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int i = 0; i < 4; ++i)
#pragma unroll
            for (int j = 0; j < 2; ++j)
#pragma unroll
                for (int kh = 0; kh < 2; ++kh)
                    acc[msub * 4 + i][nsub * 2 + j] =
                        __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                            Af[msub][i][kh], Bf[nsub][j][kh],
                            acc[msub * 4 + i][nsub * 2 + j], 0, 0, 0);
        __builtin_amdgcn_s_setprio(0);
    };

    // ---- prologue: tile0 (full) + tile1.A, then q0 prefetch -------
    const int nsteps = K / 64;
#pragma unroll
    for (int g = 0; g < 4; ++g) stageA(0, g, 0);
#pragma unroll
    for (int g = 0; g < 4; ++g) stageB(0, g, 0);
#pragma unroll
    for (int g = 0; g < 4; ++g) stageA(1, g, nsteps > 1 ? 1 : 0);
    SS_VMCNT(4);  // tile0's 8 glds landed; tile1.A (4) may fly
    SS_BAR();
    readA(0, 0);
    readB(0, 0);

    // ---- main loop: 2 K-tiles (buf0 then buf1) per iteration ------
    // Phase shape: [2 glds][same-buffer ds_read prefetch][counted
    // lgkm][16 MFMA prio1][end barrier]; P4/P8 insert vmcnt(4) + a
    // barrier before their (cross-buffer) reads.  Staged-granule
    // certification ledger (vmcnt(4) = all but last 2 phases' glds):
    //   tile 2i+1.B  issued P1        first read P4   certified P4 ✓
    //   tile 2i+2.A  issued P3,P4     first read P8   certified P8 ✓
    //   tile 2i+2.B  issued P5        first read P8   certified P8 ✓
    //   tile 2i+3.A  issued P7,P8     first read P4'  certified P4' ✓
    // Overwrite legality (glds lands ≥ issue; prior occupant's last
    // ds_read finishes ≥1 phase + barrier before each issue): P1 over
    // B(buf1) read ≤P5''; P3/P4 over A(buf0) read ≤P2; P5 over
    // B(buf0) read ≤P1; P7/P8 over A(buf1) read ≤P6.
    const int niter = K / 128;
    for (int it = 0; it < niter; ++it) {
        const int t1 = 2 * it + 1;
        const int t2 = min(2 * it + 2, nsteps - 1);
        const int t3 = min(2 * it + 3, nsteps - 1);
        // P1: q0 = (m0,n0) of buf0
        stageB(1, 0, t1);
        stageB(1, 1, t1);
        readB(0, 1);  // same-buffer prefetch: latency hides in the barrier
        SS_LGKM(4);
        mfma16(0, 0);
        SS_BAR();
        // P2: q1 = (m0,n1)  (uniform 2-glds pacing: W.B certified P4)
        stageB(1, 2, t1);
        stageB(1, 3, t1);
        readA(0, 1);
        SS_LGKM(8);
        mfma16(0, 1);
        SS_BAR();
        // P3: q2 = (m1,n0)
        stageA(0, 0, t2);
        stageA(0, 2, t2);
        SS_LGKM(0);
        mfma16(1, 0);
        SS_BAR();
        // P4: q3 = (m1,n1); certify buf1 tile; prefetch its q0 frags
        stageA(0, 1, t2);
        stageA(0, 3, t2);
        SS_VMCNT(4);
        SS_BAR();
        readA(1, 0);
        readB(1, 0);
        SS_LGKM(12);
        mfma16(1, 1);
        SS_BAR();
        // P5: q0 of buf1
        stageB(0, 0, t2);
        stageB(0, 1, t2);
        readB(1, 1);
        SS_LGKM(4);
        mfma16(0, 0);
        SS_BAR();
        // P6: q1
        stageB(0, 2, t2);
        stageB(0, 3, t2);
        readA(1, 1);
        SS_LGKM(8);
        mfma16(0, 1);
        SS_BAR();
        // P7: q2
        stageA(1, 0, t3);
        stageA(1, 2, t3);
        SS_LGKM(0);
        mfma16(1, 0);
        SS_BAR();
        // P8: q3; certify buf0's next tile; prefetch its q0 frags
        stageA(1, 1, t3);
        stageA(1, 3, t3);
        SS_VMCNT(4);
        SS_BAR();
        readA(0, 0);
        readB(0, 0);
        SS_LGKM(12);
        mfma16(1, 1);
        SS_BAR();
    }
    SS_VMCNT(0);  // drain dead prefetch glds before block exit

    // ---- epilogue: bias + ReLU + bf16 store -----------------------
#pragma unroll
    for (int I = 0; I < 8; ++I) {
#pragma unroll
        for (int J = 0; J < 4; ++J) {
            const int gcol = n0 + wn * 64 + J * 16 + lrow;
            float bv = 0.f;
            if constexpr (HAS_BIAS) bv = bf2f(bias[gcol]);
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int grow = m0 + wm * 128 + I * 16 + kch * 4 + r;
                float v = acc[I][J][r];
                if constexpr (HAS_BIAS) v += bv;
                if constexpr (RELU) v = v > 0.f ? v : 0.f;
                C[(long)grow * N + gcol] = f2bf(v);
            }
        }
    }
}

}  // namespace

// Returns false (untouched C) when the shape is outside this tier.
bool ss_gemm_nt_256(const void* A, const void* B, const void* bias, void* C,
                    int M, int N, int K, bool relu, hipStream_t stream) {
    if (M % 256 || N % 256 || K % 128 || K < 128) return false;
    // 32-bit per-lane source offsets in the kernel
    if ((long)M * K * 2 >= (1L << 31) || (long)N * K * 2 >= (1L << 31))
        return false;
    const bool has_bias = bias != nullptr;
    dim3 grid(M / 256, N / 256);
    dim3 blk(512);
    if (has_bias && relu)
        hipLaunchKernelGGL((gemm_nt_256_kernel<true, true>), grid, blk, 0,
                           stream, (const __bf16*)A, (const __bf16*)B,
                           (const __bf16*)bias, (__bf16*)C, M, N, K);
    else if (has_bias)
        hipLaunchKernelGGL((gemm_nt_256_kernel<true, false>), grid, blk, 0,
                           stream, (const __bf16*)A, (const __bf16*)B,
                           (const __bf16*)bias, (__bf16*)C, M, N, K);
    else if (relu)
        hipLaunchKernelGGL((gemm_nt_256_kernel<false, true>), grid, blk, 0,
                           stream, (const __bf16*)A, (const __bf16*)B,
                           nullptr, (__bf16*)C, M, N, K);
    else
        hipLaunchKernelGGL((gemm_nt_256_kernel<false, false>), grid, blk, 0,
                           stream, (const __bf16*)A, (const __bf16*)B,
                           nullptr, (__bf16*)C, M, N, K);
    return true;
}
