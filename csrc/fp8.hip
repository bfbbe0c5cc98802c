// MX-fp8 (OCP e4m3 + E8M0 block scales) serving tier for gfx950.
//
//   * fp8_quant_kernel: bf16 [R][K] -> u8 data [R][K] + u8 scales
//     [K/128][R][4] — one E8M0 scale per HARDWARE scale group.  The
//     gfx950 `mfma_scale_f32_16x16x128_f8f6f4` applies the scale held
//     in source lane r+16g (byte 0, opsel 0) to the INTERLEAVED
//     32-element k-set {khalf*64 + parity*16 + [0,16) ∪ [32,48)},
//     g = khalf + 2*parity — established by hardware probes
//     (scripts/probe_mx2-4.hip, profiles/r01_kernel_stats.md); the
//     quantizer's amax groups match that partition exactly.
//   * gemm_nt_f8_kernel: the gemm256.hip 8-phase skeleton at fp8:
//     same 256×256 tile, K-tile = 128 elems (= the same 128 B per
//     row, so the staging schedule and slot swizzle carry over),
//     same-phase fragment reads (lgkmcnt(0) before the MFMA burst;
//     the SIMD's partner wave hides the wait), MFMA = one scaled
//     16x16x128 op per fragment (2× the bf16 rate).  Scale rows are
//     staged by one extra glds per operand per tile at P4/P8 (after
//     their last reader's phase-end barrier), so the certification
//     checkpoints are vmcnt(6).  With EMIT_Q the epilogue emits the
//     output pre-quantized (per-row MX groups via shfl_xor over the
//     16 column lanes) — fp8-RESIDENT serving chains.
//
// This is the reference-beyond parity item (ROADMAP #6): an opt-in
// serving/inference precision mode; the bench contract stays bf16.

#include "common.h"

namespace {

typedef __attribute__((address_space(1))) const unsigned int* gptr_t;
typedef __attribute__((address_space(3))) unsigned int* lptr_t;

using i32x8 = __attribute__((ext_vector_type(8))) int;

#define SS_VMCNT(n) asm volatile("s_waitcnt vmcnt(%0)" ::"n"(n))
#define SS_LGKM(n) asm volatile("s_waitcnt lgkmcnt(%0)" ::"n"(n))
#define SS_BAR() __builtin_amdgcn_s_barrier()

// ---------------------------------------------------------------- quant

// Wave = one (row, 128-k window): lane l holds k = 2l, 2l+1 (both in
// the same hardware scale group: g = ((l>>5)&1) + 2*((l>>3)&1)); the
// per-group amax is a 4-step shfl_xor reduction over lane bits
// {0,1,2,4}; lanes with bits 0,1,2,4 clear write the 4 scale bytes.
__global__ __launch_bounds__(256) void fp8_quant_kernel(
    const __bf16* __restrict__ X,  // [R][K]
    unsigned char* __restrict__ Q,       // [R][K]
    unsigned char* __restrict__ S,       // [K/128][R][4]
    int R, int K) {
    const int nwin = K / 128;
    const long wid = ((long)blockIdx.x * blockDim.x + threadIdx.x) >> 6;
    const int lane = threadIdx.x & 63;
    if (wid >= (long)R * nwin) return;  // uniform per wave
    const int r = wid / nwin, w = wid % nwin;
    const __bf16* src = X + (long)r * K + w * 128 + 2 * lane;
    const float f0 = bf2f(src[0]), f1 = bf2f(src[1]);
    float a = fmaxf(fabsf(f0), fabsf(f1));
#pragma unroll
    for (int d : {1, 2, 4, 16}) a = fmaxf(a, __shfl_xor(a, d, 64));
    int ex = 0;
    if (a > 0.f) {
        frexpf(a / 448.f, &ex);  // a/448 = m*2^ex, m in [0.5,1)
        ex += 127;               // 2^(ex-127) >= a/448
        if (ex < 0) ex = 0;
        if (ex > 254) ex = 254;
    }
    if ((lane & 23) == 0) {  // one lane per scale group
        const int g = ((lane >> 5) & 1) + 2 * ((lane >> 3) & 1);
        S[((long)w * R + r) * 4 + g] = (unsigned char)ex;
    }
    const float inv = exp2f((float)(127 - ex));
    const int packed =
        __builtin_amdgcn_cvt_pk_fp8_f32(f0 * inv, f1 * inv, 0, false);
    *(unsigned short*)(Q + (long)r * K + w * 128 + 2 * lane) =
        (unsigned short)(packed & 0xffff);
}

// ---------------------------------------------------------------- GEMM

template <bool HAS_BIAS, bool RELU, bool EMIT_Q>
__global__ __launch_bounds__(512, 1) void gemm_nt_f8_kernel(
    const unsigned char* __restrict__ A,    // [M][K] e4m3
    const unsigned char* __restrict__ Asc,  // [K/128][M][4] e8m0
    const unsigned char* __restrict__ B,    // [N][K]
    const unsigned char* __restrict__ Bsc,  // [K/128][N][4]
    const __bf16* __restrict__ bias,        // [N]
    __bf16* __restrict__ C,                 // [M][N] bf16 (EMIT_Q==0)
    unsigned char* __restrict__ Cq,         // [M][N] e4m3 (EMIT_Q==1)
    unsigned char* __restrict__ Cs,         // [N/128][M][4]
    int M, int N, int K) {
    // [op][buf][256 rows × 128 k data, then 256×4 scale rows] u8 —
    // ONE LDS object (a second __shared__ written by glds makes the
    // compiler order ds_reads against the DMA with vmcnt(0))
    __shared__ unsigned char lds[2][2][256 * 128 + 256 * 4];

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wave = tid >> 6;
    const int wm = wave >> 2;
    const int wn = wave & 3;

    const int gx = gridDim.x, gy = gridDim.y;
    const int nwg = gx * gy;
    const int hw = blockIdx.x + gx * blockIdx.y;
    const int xcd = hw % 8, q8 = nwg / 8, r8 = nwg % 8;
    const int wid = (xcd < r8 ? xcd * (q8 + 1)
                              : r8 * (q8 + 1) + (xcd - r8) * q8) + hw / 8;
    const int m0 = ((wid / gy) % gx) * 256;
    const int n0 = (wid % gy) * 256;

    // ---- staging (identical arithmetic to gemm256.hip: a row is
    // 128 B; granule = 64 rows = 1 glds/wave; slot swizzle
    // s ^ ((row>>1)&7) is conflict-free for the b128 fragment reads
    // here too: lanes 0-15 read slot (l>>4)*2+h at 16 distinct
    // bank-quads after the XOR) --------------------------------------
    const int lrow8 = lane >> 3;
    const int sig = lane & 7;
    const int slog = sig ^ (((wave & 1) << 2) | (lrow8 >> 1));
    const int avoff = (m0 + wave * 8 + lrow8) * K + slog * 16;
    const int bvoff = (n0 + wave * 8 + lrow8) * K + slog * 16;
    const int gstride = 64 * K;

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
    // scale rows: 256 rows × 4 B = 1 KiB = one glds; issued by EVERY
    // wave (identical redundant writes keep the per-wave vmcnt ledger
    // uniform; same-value overlap is benign)
    auto stageS = [&](int op, int buf, int t) {
        const unsigned char* src =
            op ? Bsc + ((long)t * N + n0) * 4 : Asc + ((long)t * M + m0) * 4;
        __builtin_amdgcn_global_load_lds(
            (gptr_t)(src + lane * 16),
            (lptr_t)&lds[op][buf][256 * 128], 16, 0, 0);
    };

    // fragment reads: lane l reads A[row][ (l>>4)*32 + h*16 .. +16 )
    const int lrow = lane & 15;
    const int kch = lane >> 4;
    // slot bit0 (the 16-B half) maps to address bit 4 after the XOR
    // swizzle, so the +1 half is base ^ 16 — two pointers total.
    const int aoffL = (wm * 128 + lrow) * 128 + ((kch * 2) ^ (lrow >> 1)) * 16;
    const int boffL = (wn * 64 + lrow) * 128 + ((kch * 2) ^ (lrow >> 1)) * 16;
    using i32x4 = __attribute__((ext_vector_type(4))) int;
    // SINGLE A set (same-phase reads, guide-style lgkmcnt(0) before
    // MFMA: the SIMD's partner wave fills the read-wait), double B
    // set (reused across the two msub phases).  Elementwise fills —
    // address-taking a register array spills it to scratch.
    i32x8 Af[4];     // current msub's 4 fragments
    i32x8 Bf[2][2];  // [nsub][j]
    unsigned int SaP;  // 4 scale bytes of current msub's rows
    unsigned int SbP;  // 4 scale bytes (nsub*2+j)

    auto readA = [&](int buf, int msub) {
#pragma unroll
        for (int i = 0; i < 4; ++i) {
            const int off = buf * 33792  /* data+scale stride */ + (msub * 64 + i * 16) * 128;
            i32x4 lo = *(const i32x4*)((const char*)&lds[0][0][0] +
                                       (aoffL + off));
            i32x4 hi = *(const i32x4*)((const char*)&lds[0][0][0] +
                                       ((aoffL ^ 16) + off));
#pragma unroll
            for (int e = 0; e < 4; ++e) {
                Af[i][e] = lo[e];
                Af[i][e + 4] = hi[e];
            }
        }
    };
    auto readB = [&](int buf, int nsub) {
#pragma unroll
        for (int j = 0; j < 2; ++j) {
            const int off = buf * 33792  /* data+scale stride */ + (nsub * 32 + j * 16) * 128;
            i32x4 lo = *(const i32x4*)((const char*)&lds[1][0][0] +
                                       (boffL + off));
            i32x4 hi = *(const i32x4*)((const char*)&lds[1][0][0] +
                                       ((boffL ^ 16) + off));
#pragma unroll
            for (int e = 0; e < 4; ++e) {
                Bf[nsub][j][e] = lo[e];
                Bf[nsub][j][e + 4] = hi[e];
            }
        }
    };
    // lane's scale byte = byte (l>>4) of the row's 4 — read as u8
    auto readSA = [&](int buf, int msub) {
        unsigned int p = 0;
#pragma unroll
        for (int i = 0; i < 4; ++i) {
            const int r = wm * 128 + msub * 64 + i * 16 + lrow;
            p |= (unsigned int)lds[0][buf][256 * 128 + r * 4 + kch]
                 << (8 * i);
        }
        SaP = p;
    };
    auto readSB = [&](int buf) {
        unsigned int q = 0;
#pragma unroll
        for (int nsub = 0; nsub < 2; ++nsub)
#pragma unroll
            for (int j = 0; j < 2; ++j) {
                const int r = wn * 64 + nsub * 32 + j * 16 + lrow;
                q |= (unsigned int)lds[1][buf][256 * 128 + r * 4 + kch]
                     << (8 * (nsub * 2 + j));
            }
        SbP = q;
    };

    f32x4 acc[8][4];
#pragma unroll
    for (int i = 0; i < 8; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

    auto mfma8 = [&](int msub, int nsub) {
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int i = 0; i < 4; ++i)
#pragma unroll
            for (int j = 0; j < 2; ++j)
                acc[msub * 4 + i][nsub * 2 + j] =
                    __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
                        Af[i], Bf[nsub][j],
                        acc[msub * 4 + i][nsub * 2 + j], 0, 0,
                        0, (int)((SaP >> (8 * i)) & 0xff),
                        0, (int)((SbP >> (8 * (nsub * 2 + j))) & 0xff));
        __builtin_amdgcn_s_setprio(0);
    };

    // ---- prologue -------------------------------------------------
    const int nsteps = K / 128;
#pragma unroll
    for (int g = 0; g < 4; ++g) stageA(0, g, 0);
#pragma unroll
    for (int g = 0; g < 4; ++g) stageB(0, g, 0);
    stageS(0, 0, 0);
    stageS(1, 0, 0);
    stageS(0, 1, nsteps > 1 ? 1 : 0);  // tile1 scales (read at P4!)
    stageS(1, 1, nsteps > 1 ? 1 : 0);
#pragma unroll
    for (int g = 0; g < 4; ++g) stageA(1, g, nsteps > 1 ? 1 : 0);
    SS_VMCNT(6);  // tile0 data+scales landed; tile1 scales+A may fly
    SS_BAR();

    // ---- main loop: same-phase reads (guide-style lgkmcnt(0) before
    // MFMA — the SIMD partner wave covers the read wait); staging
    // schedule and vmcnt(6) certification as derived for gemm256 +
    // scale rows at P4/P8.  Reads at P1/P5 touch the buffer certified
    // by the PREVIOUS vmcnt barrier.
    const int niter = nsteps / 2;
    for (int it = 0; it < niter; ++it) {
        const int t1 = 2 * it + 1;
        const int t2 = min(2 * it + 2, nsteps - 1);
        const int t3 = min(2 * it + 3, nsteps - 1);
        // P1: q0 of buf0
        readA(0, 0);
        readB(0, 0);
        readSA(0, 0);
        readSB(0);
        stageB(1, 0, t1);
        stageB(1, 1, t1);
        SS_LGKM(0);
        mfma8(0, 0);
        SS_BAR();
        // P2: q1
        readB(0, 1);
        stageB(1, 2, t1);
        stageB(1, 3, t1);
        SS_LGKM(0);
        mfma8(0, 1);
        SS_BAR();
        // P3: q2
        readA(0, 1);
        readSA(0, 1);
        stageA(0, 0, t2);
        stageA(0, 2, t2);
        SS_LGKM(0);
        mfma8(1, 0);
        SS_BAR();
        // P4: q3 (+ X scale rows — bufA scale region last read at
        // P3, freed by its end barrier); then certify buf1
        stageA(0, 1, t2);
        stageA(0, 3, t2);
        stageS(0, 0, t2);
        stageS(1, 0, t2);
        SS_LGKM(0);
        mfma8(1, 1);
        SS_VMCNT(6);
        SS_BAR();
        // P5: q0 of buf1 (reads legal: buf1 certified at P4's barrier)
        readA(1, 0);
        readB(1, 0);
        readSA(1, 0);
        readSB(1);
        stageB(0, 0, t2);
        stageB(0, 1, t2);
        SS_LGKM(0);
        mfma8(0, 0);
        SS_BAR();
        // P6: q1
        readB(1, 1);
        stageB(0, 2, t2);
        stageB(0, 3, t2);
        SS_LGKM(0);
        mfma8(0, 1);
        SS_BAR();
        // P7: q2
        readA(1, 1);
        readSA(1, 1);
        stageA(1, 0, t3);
        stageA(1, 2, t3);
        SS_LGKM(0);
        mfma8(1, 0);
        SS_BAR();
        // P8: q3 (+ Y scale rows, freed by P7's barrier); certify
        // buf0's next tile
        stageA(1, 1, t3);
        stageA(1, 3, t3);
        stageS(0, 1, t3);
        stageS(1, 1, t3);
        SS_LGKM(0);
        mfma8(1, 1);
        SS_VMCNT(6);
        SS_BAR();
    }
    SS_VMCNT(0);

    // ---- epilogue -------------------------------------------------
    // EMIT_Q: the output is quantized IN the epilogue (fused — no
    // separate pass, no bf16 round trip): the wave's 64 columns fall
    // in ONE 128-col scale window (khalf = wn&1) and parity = J&1,
    // so each (row, parity) MX group amax is a 4-step shfl_xor
    // reduction over the 16 column lanes — no LDS, no cross-wave
    // sync.  Scale/data layouts match this kernel's own INPUT format,
    // so serving chains stay fp8-resident end to end.
#pragma unroll
    for (int I = 0; I < 8; ++I) {
        float bv[4];
#pragma unroll
        for (int J = 0; J < 4; ++J) {
            bv[J] = 0.f;
            if constexpr (HAS_BIAS)
                bv[J] = bf2f(bias[n0 + wn * 64 + J * 16 + lrow]);
        }
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            const int grow = m0 + wm * 128 + I * 16 + kch * 4 + r;
            float v[4];
#pragma unroll
            for (int J = 0; J < 4; ++J) {
                v[J] = acc[I][J][r];
                if constexpr (HAS_BIAS) v[J] += bv[J];
                if constexpr (RELU) v[J] = v[J] > 0.f ? v[J] : 0.f;
            }
            if constexpr (!EMIT_Q) {
#pragma unroll
                for (int J = 0; J < 4; ++J)
                    C[(long)grow * N + n0 + wn * 64 + J * 16 + lrow] =
                        f2bf(v[J]);
            } else {
                // group amax per parity (J&1) over J-pair + 16 lanes
                float m0p = fmaxf(fabsf(v[0]), fabsf(v[2]));
                float m1p = fmaxf(fabsf(v[1]), fabsf(v[3]));
#pragma unroll
                for (int d : {1, 2, 4, 8}) {
                    m0p = fmaxf(m0p, __shfl_xor(m0p, d, 64));
                    m1p = fmaxf(m1p, __shfl_xor(m1p, d, 64));
                }
                int e0 = 0, e1 = 0;
                if (m0p > 0.f) {
                    frexpf(m0p / 448.f, &e0);
                    e0 = min(max(e0 + 127, 0), 254);
                }
                if (m1p > 0.f) {
                    frexpf(m1p / 448.f, &e1);
                    e1 = min(max(e1 + 127, 0), 254);
                }
                const int w = (n0 + wn * 64) >> 7;
                if (lrow == 0) {
                    Cs[((long)w * M + grow) * 4 + ((wn & 1) + 0)] =
                        (unsigned char)e0;
                    Cs[((long)w * M + grow) * 4 + ((wn & 1) + 2)] =
                        (unsigned char)e1;
                }
                const float i0 = exp2f((float)(127 - e0));
                const float i1 = exp2f((float)(127 - e1));
                const int p02 = __builtin_amdgcn_cvt_pk_fp8_f32(
                    v[0] * i0, v[2] * i0, 0, false);
                const int p13 = __builtin_amdgcn_cvt_pk_fp8_f32(
                    v[1] * i1, v[3] * i1, 0, false);
                const long rb = (long)grow * N + n0 + wn * 64 + lrow;
                Cq[rb] = (unsigned char)(p02 & 0xff);
                Cq[rb + 32] = (unsigned char)((p02 >> 8) & 0xff);
                Cq[rb + 16] = (unsigned char)(p13 & 0xff);
                Cq[rb + 48] = (unsigned char)((p13 >> 8) & 0xff);
            }
        }
    }
}

}  // namespace

void ss_fp8_quantize(const void* X, void* Q, void* S, int R, int K,
                     hipStream_t stream) {
    const long total = (long)R * (K / 128) * 64;  // one wave per window
    dim3 grid((total + 255) / 256);
    dim3 blk(256);
    hipLaunchKernelGGL(fp8_quant_kernel, grid, blk, 0, stream,
                       (const __bf16*)X, (unsigned char*)Q,
                       (unsigned char*)S, R, K);
}

bool ss_gemm_nt_f8(const void* A, const void* Asc, const void* B,
                   const void* Bsc, const void* bias, void* C, int M, int N,
                   int K, bool relu, hipStream_t stream) {
    if (M % 256 || N % 256 || K % 256 || K < 256) return false;
    if ((long)M * K >= (1L << 31) || (long)N * K >= (1L << 31)) return false;
    dim3 grid(M / 256, N / 256);
    dim3 blk(512);
    const bool hb = bias != nullptr;
#define F8LAUNCH(HB, RL)                                                     \
    hipLaunchKernelGGL((gemm_nt_f8_kernel<HB, RL, false>), grid, blk, 0,     \
                       stream, (const unsigned char*)A,                      \
                       (const unsigned char*)Asc, (const unsigned char*)B,   \
                       (const unsigned char*)Bsc, (const __bf16*)bias,       \
                       (__bf16*)C, nullptr, nullptr, M, N, K)
    if (hb && relu) F8LAUNCH(true, true);
    else if (hb) F8LAUNCH(true, false);
    else if (relu) F8LAUNCH(false, true);
    else F8LAUNCH(false, false);
#undef F8LAUNCH
    return true;
}

// Fused-quant output: C is emitted as (e4m3 data, E8M0 scales) in the
// same layout this kernel consumes — fp8-resident serving chains.
bool ss_gemm_nt_f8_q(const void* A, const void* Asc, const void* B,
                     const void* Bsc, const void* bias, void* Cq, void* Cs,
                     int M, int N, int K, bool relu, hipStream_t stream) {
    if (M % 256 || N % 256 || K % 256 || K < 256) return false;
    if ((long)M * K >= (1L << 31) || (long)N * K >= (1L << 31)) return false;
    dim3 grid(M / 256, N / 256);
    dim3 blk(512);
    const bool hb = bias != nullptr;
#define F8QLAUNCH(HB, RL)                                                    \
    hipLaunchKernelGGL((gemm_nt_f8_kernel<HB, RL, true>), grid, blk, 0,      \
                       stream, (const unsigned char*)A,                      \
                       (const unsigned char*)Asc, (const unsigned char*)B,   \
                       (const unsigned char*)Bsc, (const __bf16*)bias,       \
                       nullptr, (unsigned char*)Cq, (unsigned char*)Cs, M,   \
                       N, K)
    if (hb && relu) F8QLAUNCH(true, true);
    else if (hb) F8QLAUNCH(true, false);
    else if (relu) F8QLAUNCH(false, true);
    else F8QLAUNCH(false, false);
#undef F8QLAUNCH
    return true;
}
