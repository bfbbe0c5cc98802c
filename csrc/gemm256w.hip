// 256×256-tile 8-phase MFMA bf16 GEMM, 4-wave / 128×128-wave-tile
// variant (gfx950 / MI355X) — the LDS-read-bandwidth ceiling breaker.
//
//   C[M,N] = A[M,K] · B[N,K]^T (+bias) (+ReLU)   M%256==0, N%256==0,
//                                                K%128==0, K>=128.
//
// Why a second structure next to gemm256.hip's 8-wave kernel: LDS
// fragment-read traffic per K-tile per CU is (wm+wn)·32 KiB for a
// wm×wn wave grid covering the 256² tile.  The 8-wave 2×4 grid reads
// 192 KiB = 768 cyc at 256 B/clk against 512 cyc of MFMA issue —
// capping ANY such kernel at ~67% MFMA util (~1670 TF, the measured
// v1 ceiling; docs/KERNELS.md).  A 2×2 grid of 128×128 wave tiles
// reads 128 KiB = 512 cyc — LDS and MFMA issue balance exactly, so
// the structural ceiling moves to the MFMA peak.  The cost: 256
// accumulator VGPRs per lane (acc[8][8]·f32x4) + 128 fragment VGPRs
// → 1 wave/SIMD, so ALL latency hiding comes from the static 8-phase
// pipeline (3-phase staging headroom, 1-phase-ahead fragment reads),
// not from co-resident waves.
//
// Everything else carries over from gemm256.hip verbatim: LDS images
// and the s^((row>>1)&7) slot swizzle, the staging call schedule
// (P1,P2: t1.B | P3,P4: t2.A | P5,P6: t2.B | P7,P8: t3.A), the
// overwrite/freeing ledger, and the certification points — with every
// per-wave glds doubled (4 waves must move the same bytes), so the
// certification waits become vmcnt(8) (= "all but the last two
// phases' glds landed").  Each phase runs 32 MFMA (a 64×64×64
// quadrant) under setprio(1).
//
// Reference scope: linear forward/dgrad (functional.py:13-21) at wide
// shapes, same as gemm256.hip.

#include "common.h"

namespace {

typedef __attribute__((address_space(1))) const unsigned int* gptr_t;
typedef __attribute__((address_space(3))) unsigned int* lptr_t;

#define SS_VMCNT(n) asm volatile("s_waitcnt vmcnt(%0)" ::"n"(n))
#define SS_LGKM(n) asm volatile("s_waitcnt lgkmcnt(%0)" ::"n"(n))
#define SS_BAR() __builtin_amdgcn_s_barrier()

template <bool HAS_BIAS, bool RELU>
__global__ __launch_bounds__(256, 1) void gemm_nt_256w_kernel(
    const __bf16* __restrict__ A,     // [M][K]
    const __bf16* __restrict__ B,     // [N][K]
    const __bf16* __restrict__ bias,  // [N]
    __bf16* __restrict__ C,           // [M][N]
    int M, int N, int K) {
    __shared__ ushort lds[2][2][256 * 64];  // 128 KiB, same as v1

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wave = tid >> 6;  // 0..3
    const int wm = wave >> 1;   // 2 M-halves
    const int wn = wave & 1;    // 2 N-halves

    // XCD-aware bijective block remap (identical to v1)
    const int gx = gridDim.x, gy = gridDim.y;
    const int nwg = gx * gy;
    const int hw = blockIdx.x + gx * blockIdx.y;
    const int xcd = hw % 8, q8 = nwg / 8, r8 = nwg % 8;
    const int wid = (xcd < r8 ? xcd * (q8 + 1)
                              : r8 * (q8 + 1) + (xcd - r8) * q8) + hw / 8;
    const int m0 = ((wid / gy) % gx) * 256;
    const int n0 = (wid % gy) * 256;

    // ---- glds staging: wave w stages rows [g*64 + w*16, +16) of
    // granule g as TWO 8-row glds (v1 staged 8 rows with 8 waves).
    // Slot pre-swizzle: row_in_tile = g*64 + wave*16 + h*8 + lrow8;
    // (row>>1)&7 sees (wave*16 + h*8 + lrow8)>>1 & 7 = wave*8+h*4+
    // (lrow8>>1) — wave/h/lane bits only (g*64 ≡ 0 mod 16 rows).
    const int lrow8 = lane >> 3;
    const int sig = lane & 7;
    auto slog = [&](int h) {
        return sig ^ ((((wave * 16 + h * 8 + lrow8) >> 1) & 7));
    };
    const int avoff0 = (m0 + wave * 16 + 0 * 8 + lrow8) * K * 2 + slog(0) * 16;
    const int avoff1 = (m0 + wave * 16 + 1 * 8 + lrow8) * K * 2 + slog(1) * 16;
    const int bvoff0 = (n0 + wave * 16 + 0 * 8 + lrow8) * K * 2 + slog(0) * 16;
    const int bvoff1 = (n0 + wave * 16 + 1 * 8 + lrow8) * K * 2 + slog(1) * 16;
    const int gstride = 64 * K * 2;  // granule row stride (bytes)

    auto stageA = [&](int buf, int g, int t) {  // 2 glds
        __builtin_amdgcn_global_load_lds(
            (gptr_t)((const char*)A + (avoff0 + g * gstride + t * 128)),
            (lptr_t)((char*)&lds[0][buf][0] + (g * 64 + wave * 16) * 128),
            16, 0, 0);
        __builtin_amdgcn_global_load_lds(
            (gptr_t)((const char*)A + (avoff1 + g * gstride + t * 128)),
            (lptr_t)((char*)&lds[0][buf][0] + (g * 64 + wave * 16 + 8) * 128),
            16, 0, 0);
    };
    auto stageB = [&](int buf, int g, int t) {
        __builtin_amdgcn_global_load_lds(
            (gptr_t)((const char*)B + (bvoff0 + g * gstride + t * 128)),
            (lptr_t)((char*)&lds[1][buf][0] + (g * 64 + wave * 16) * 128),
            16, 0, 0);
        __builtin_amdgcn_global_load_lds(
            (gptr_t)((const char*)B + (bvoff1 + g * gstride + t * 128)),
            (lptr_t)((char*)&lds[1][buf][0] + (g * 64 + wave * 16 + 8) * 128),
            16, 0, 0);
    };

    // ---- fragment reads (ds_read_b128, swizzled; same as v1) -------
    const int lrow = lane & 15;
    const int kch = lane >> 4;
    const int phys0 = (kch ^ (lrow >> 1)) * 16;
    const char* rdA0 = (const char*)&lds[0][0][0] + (wm * 128 + lrow) * 128 + phys0;
    const char* rdA1 = (const char*)&lds[0][0][0] + (wm * 128 + lrow) * 128 + (phys0 ^ 64);
    const char* rdB0 = (const char*)&lds[1][0][0] + (wn * 128 + lrow) * 128 + phys0;
    const char* rdB1 = (const char*)&lds[1][0][0] + (wn * 128 + lrow) * 128 + (phys0 ^ 64);
    bf16x8 Af[2][4][2];  // [msub][i][kh] — the wave's 128 A rows
    bf16x8 Bf[2][4][2];  // [nsub][j][kh] — the wave's 128 B rows

    auto readA = [&](int buf, int msub) {  // 8 ds_read_b128
#pragma unroll
        for (int i = 0; i < 4; ++i) {
            const int off = buf * 32768 + (msub * 64 + i * 16) * 128;
            Af[msub][i][0] = *(const bf16x8*)(rdA0 + off);
            Af[msub][i][1] = *(const bf16x8*)(rdA1 + off);
        }
    };
    auto readB = [&](int buf, int nsub) {  // 8 ds_read_b128
#pragma unroll
        for (int j = 0; j < 4; ++j) {
            const int off = buf * 32768 + (nsub * 64 + j * 16) * 128;
            Bf[nsub][j][0] = *(const bf16x8*)(rdB0 + off);
            Bf[nsub][j][1] = *(const bf16x8*)(rdB1 + off);
        }
    };

    f32x4 acc[8][8];
#pragma unroll
    for (int i = 0; i < 8; ++i)
#pragma unroll
        for (int j = 0; j < 8; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

    auto mfma32 = [&](int msub, int nsub) {  // one 64×64×64 quadrant
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int i = 0; i < 4; ++i)
#pragma unroll
            for (int j = 0; j < 4; ++j)
#pragma unroll
                for (int kh = 0; kh < 2; ++kh)
                    acc[msub * 4 + i][nsub * 4 + j] =
                        __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                            Af[msub][i][kh], Bf[nsub][j][kh],
                            acc[msub * 4 + i][nsub * 4 + j], 0, 0, 0);
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
    SS_VMCNT(8);  // tile0's 16 glds landed; tile1.A (8) may fly
    SS_BAR();
    readA(0, 0);
    readB(0, 0);

    // ---- main loop: identical call/cert schedule to v1, glds and
    // read counts doubled.  Certification ledger (vmcnt(8) = all but
    // the last two phases' 4-glds groups landed): same tile→phase map
    // as v1 (see gemm256.hip P1..P8 ledger comments).
    const int niter = K / 128;
    for (int it = 0; it < niter; ++it) {
        const int t1 = 2 * it + 1;
        const int t2 = min(2 * it + 2, nsteps - 1);
        const int t3 = min(2 * it + 3, nsteps - 1);
        // P1: q(0,0) of buf0
        stageB(1, 0, t1);
        stageB(1, 1, t1);
        readB(0, 1);  // prefetch B half 1
        SS_LGKM(8);
        mfma32(0, 0);
        SS_BAR();
        // P2: q(0,1)
        stageB(1, 2, t1);
        stageB(1, 3, t1);
        readA(0, 1);  // prefetch A half 1
        SS_LGKM(8);
        mfma32(0, 1);
        SS_BAR();
        // P3: q(1,0)
        stageA(0, 0, t2);
        stageA(0, 2, t2);
        SS_LGKM(0);
        mfma32(1, 0);
        SS_BAR();
        // P4: q(1,1); certify buf1 tile; prefetch its q0 frags
        stageA(0, 1, t2);
        stageA(0, 3, t2);
        SS_VMCNT(8);
        SS_BAR();
        readA(1, 0);
        SS_LGKM(15);  // ISA cap: waits 1 of our own 8, all priors done
        mfma32(1, 1);
        readB(1, 0);  // consumed P5, covered by its LGKM(8)
        SS_BAR();
        // P5: q(0,0) of buf1
        stageB(0, 0, t2);
        stageB(0, 1, t2);
        readB(1, 1);
        SS_LGKM(8);
        mfma32(0, 0);
        SS_BAR();
        // P6: q(0,1)
        stageB(0, 2, t2);
        stageB(0, 3, t2);
        readA(1, 1);
        SS_LGKM(8);
        mfma32(0, 1);
        SS_BAR();
        // P7: q(1,0)
        stageA(1, 0, t3);
        stageA(1, 2, t3);
        SS_LGKM(0);
        mfma32(1, 0);
        SS_BAR();
        // P8: q(1,1); certify buf0's next tile; prefetch its q0 frags
        stageA(1, 1, t3);
        stageA(1, 3, t3);
        SS_VMCNT(8);
        SS_BAR();
        readA(0, 0);
        SS_LGKM(15);
        mfma32(1, 1);
        readB(0, 0);
        SS_BAR();
    }
    SS_VMCNT(0);  // drain dead prefetch glds before block exit

    // ---- epilogue: bias + ReLU + bf16 store -----------------------
#pragma unroll
    for (int I = 0; I < 8; ++I) {
#pragma unroll
        for (int J = 0; J < 8; ++J) {
            const int gcol = n0 + wn * 128 + J * 16 + lrow;
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
bool ss_gemm_nt_256w(const void* A, const void* B, const void* bias, void* C,
                     int M, int N, int K, bool relu, hipStream_t stream) {
    if (M % 256 || N % 256 || K % 128 || K < 128) return false;
    if ((long)M * K * 2 >= (1L << 31) || (long)N * K * 2 >= (1L << 31))
        return false;
    const bool has_bias = bias != nullptr;
    dim3 grid(M / 256, N / 256);
    dim3 blk(256);
    if (has_bias && relu)
        hipLaunchKernelGGL((gemm_nt_256w_kernel<true, true>), grid, blk, 0,
                           stream, (const __bf16*)A, (const __bf16*)B,
                           (const __bf16*)bias, (__bf16*)C, M, N, K);
    else if (has_bias)
        hipLaunchKernelGGL((gemm_nt_256w_kernel<true, false>), grid, blk, 0,
                           stream, (const __bf16*)A, (const __bf16*)B,
                           (const __bf16*)bias, (__bf16*)C, M, N, K);
    else if (relu)
        hipLaunchKernelGGL((gemm_nt_256w_kernel<false, true>), grid, blk, 0,
                           stream, (const __bf16*)A, (const __bf16*)B,
                           nullptr, (__bf16*)C, M, N, K);
    else
        hipLaunchKernelGGL((gemm_nt_256w_kernel<false, false>), grid, blk, 0,
                           stream, (const __bf16*)A, (const __bf16*)B,
                           nullptr, (__bf16*)C, M, N, K);
    return true;
}
