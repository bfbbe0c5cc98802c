// 256×256-tile 8-phase TN weight-gradient GEMM (gfx950 / MI355X).
//
//   gW[Mo,N] += dY[Kb,Mo]^T · X[Kb,N]     Mo%256==0, N%256==0,
//                                         Kb%128==0 (per split).
//
// The 8-phase skeleton of gemm256.hip carried to the transpose-read
// regime of wgrad (contraction over batch ROWS):
//   * LDS images are BLOCKED [mblk][ksub][4][16] bf16 with subtile
//     position p = (s & ~3) | swap01(s & 3) — swapping ksub bits 0,1
//     places the four ds_read_b64_tr_b16 lane-groups of a wave on
//     {row even|odd} × {phase 0|1} of the 256-B bank rows, i.e. the
//     optimal 2-clk pattern with NO padding (the 144-B-padded image
//     of wgrad_tn_kernel leaves a 4-deep overlap on banks 24-31 —
//     the residual "PMC 320/wave" conflicts).  Unpadded, a granule
//     (4 mblks = 64 output rows × all 64 k) is exactly 8 KiB = one
//     lane-linear glds per wave, so the gemm256 staging schedule
//     {P1,P2: W.B | P3,P4: X.A | P5,P6: X.B | P7,P8: Y.A} with
//     vmcnt(4) at P4/P8 transfers verbatim (same freeing proof).
//   * glds performs the transpose: lane piece r of subtile s maps to
//     global (k = ksub*4 + (r>>1), m = mblk*16 + (r&1)*8) — a
//     per-lane source scatter, lane-linear LDS landing.
//   * fragments come out via the gfx950 hardware transpose read
//     (__builtin_amdgcn_ds_read_tr16_b64_v4bf16), one phase ahead of
//     their MFMA consumer, counted lgkmcnt (>15-read phases split
//     around the MFMA block: the ISA field caps at 15).
//   * f32 atomicAdd epilogue — the accumulation IS the reference's
//     `grad +=` (layers.py:135-136); bias grad is NOT fused here
//     (the launcher pairs this kernel with colsum_kernel).
//
// Measured vs wgrad_tn_kernel<128,128> at 4096²×16384: see
// scripts/test_gemm256.py --wgrad / profiles/r01_kernel_stats.md.

#include "common.h"

namespace {

typedef __attribute__((address_space(1))) const unsigned int* gptr_t;
typedef __attribute__((address_space(3))) unsigned int* lptr_t;

using bf16x4v = __attribute__((ext_vector_type(4))) __bf16;
typedef __attribute__((address_space(3))) bf16x4v* lds_v4p;

#define SS_VMCNT(n) asm volatile("s_waitcnt vmcnt(%0)" ::"n"(n))

// Raw-asm transpose read: the ds_read_tr INTRINSIC's LDS-read memory
// effect makes SIInsertWaitcnts order it against in-flight LDS-DMA
// with a pipeline-killing vmcnt(0) per phase; the asm form leaves
// ordering to this kernel's counted s_waitcnt (which is exact).
__device__ __forceinline__ bf16x4v tr_read(lds_v4p p) {
    bf16x4v out;
    asm volatile("ds_read_b64_tr_b16 %0, %1" : "=v"(out) : "v"(p));
    return out;
}
#define SS_LGKM(n) asm volatile("s_waitcnt lgkmcnt(%0)" ::"n"(n))
#define SS_BAR() __builtin_amdgcn_s_barrier()

__device__ __forceinline__ int swap01(int x) {
    return (x & ~3) | ((x & 1) << 1) | ((x >> 1) & 1);
}

__global__ __launch_bounds__(512, 1) void wgrad_tn_256_kernel(
    const __bf16* __restrict__ dY,  // [Kb][Mo]
    const __bf16* __restrict__ X,   // [Kb][N]
    float* __restrict__ gW,         // [Mo][N] += via atomics
    int Mo, int N, int Kb, int k_per_split) {
    // [op dY=0/X=1][buf][16 mblk][16 ksub][4][16] bf16 = 128 KiB
    __shared__ ushort lds[2][2][16 * 16 * 64];

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wave = tid >> 6;  // 0..7
    const int wm = wave >> 2;   // 2 M-halves (output rows = dY cols)
    const int wn = wave & 3;    // 4 N-quarters (output cols = X cols)

    // XCD-aware bijective remap over (x=m-tiles, y=n-tiles, z=splits)
    const int gx = gridDim.x, gy = gridDim.y;
    const int nwg = gx * gy * gridDim.z;
    const int hw = blockIdx.x + gx * (blockIdx.y + gy * blockIdx.z);
    const int xcd = hw % 8, q8 = nwg / 8, r8 = nwg % 8;
    const int wid = (xcd < r8 ? xcd * (q8 + 1)
                              : r8 * (q8 + 1) + (xcd - r8) * q8) + hw / 8;
    const int m0 = ((wid / gy) % gx) * 256;
    const int n0 = (wid % gy) * 256;
    const int kbeg = (wid / (gx * gy)) * k_per_split;
    const int kend = min(Kb, kbeg + k_per_split);
    const int ksteps = (kend - kbeg) / 64;  // multiple of 2

    // ---- glds staging constants ----------------------------------
    // Wave w stages piece run [g*512 + w*64, +64) of a granule; lane
    // covers position p = g*64 + w*8 + (lane>>3), piece r = lane&7.
    const int p_l = wave * 8 + (lane >> 3);
    const int s_l = swap01(p_l);          // subtile (g-invariant bits)
    const int ksub_l = s_l & 15;
    const int mrel_l = s_l >> 4;          // 0..3 within the granule
    const int k_l = ksub_l * 4 + ((lane & 7) >> 1);
    const int m_l = mrel_l * 16 + (lane & 1) * 8;
    // 32-bit per-lane byte offsets (launcher guards Kb·ld·2 < 2³¹)
    const int avoff = (k_l * Mo + m0 + m_l) * 2;
    const int bvoff = (k_l * N + n0 + m_l) * 2;

    auto stageA = [&](int buf, int g, int kt) {  // granule g of dY tile kt
        __builtin_amdgcn_global_load_lds(
            (gptr_t)((const char*)dY + (avoff + ((kbeg + kt * 64) * Mo + g * 64) * 2)),
            (lptr_t)((char*)&lds[0][buf][0] + g * 8192 + wave * 1024),
            16, 0, 0);
    };
    auto stageB = [&](int buf, int g, int kt) {
        __builtin_amdgcn_global_load_lds(
            (gptr_t)((const char*)X + (bvoff + ((kbeg + kt * 64) * N + g * 64) * 2)),
            (lptr_t)((char*)&lds[1][buf][0] + g * 8192 + wave * 1024),
            16, 0, 0);
    };

    // ---- hardware-transpose fragment reads ------------------------
    // Read (mblk, kk, hi): subtile s = mblk*16 + kk*8 + kch*2 (+1 for
    // hi) → position base + immediates; the lane-variant part
    // ((kch>>1)*512 + (kch&1)*128 + lrow*8 bytes) folds into ONE base
    // pointer per operand.
    const int lrow = lane & 15;
    const int kch = lane >> 4;
    const char* rdA = (const char*)&lds[0][0][0] +
                      ((kch >> 1) * 512 + (kch & 1) * 128) + lrow * 8;
    const char* rdB = (const char*)&lds[1][0][0] +
                      ((kch >> 1) * 512 + (kch & 1) * 128) + lrow * 8;
    bf16x8 Af[2][4][2];  // [msub][i][kk]
    bf16x8 Bf[2][2][2];  // [nsub][j][kk]

    auto readAfrag = [&](int buf, int msub, int i, int kk) {
        const int off = buf * 32768 + (wm * 8 + msub * 4 + i) * 2048 + kk * 1024;
        bf16x4v lo = tr_read((lds_v4p)(rdA + off));
        bf16x4v hi = tr_read((lds_v4p)(rdA + off + 256));
#pragma unroll
        for (int e = 0; e < 4; ++e) {
            Af[msub][i][kk][e] = lo[e];
            Af[msub][i][kk][e + 4] = hi[e];
        }
    };
    auto readBfrag = [&](int buf, int nsub, int j, int kk) {
        const int off = buf * 32768 + (wn * 4 + nsub * 2 + j) * 2048 + kk * 1024;
        bf16x4v lo = tr_read((lds_v4p)(rdB + off));
        bf16x4v hi = tr_read((lds_v4p)(rdB + off + 256));
#pragma unroll
        for (int e = 0; e < 4; ++e) {
            Bf[nsub][j][kk][e] = lo[e];
            Bf[nsub][j][kk][e + 4] = hi[e];
        }
    };
    auto readA = [&](int buf, int msub) {  // 8 tr-read pairs
#pragma unroll
        for (int i = 0; i < 4; ++i)
#pragma unroll
            for (int kk = 0; kk < 2; ++kk) readAfrag(buf, msub, i, kk);
    };
    auto readAhalf = [&](int buf, int msub, int kk) {
#pragma unroll
        for (int i = 0; i < 4; ++i) readAfrag(buf, msub, i, kk);
    };
    auto readB = [&](int buf, int nsub) {
#pragma unroll
        for (int j = 0; j < 2; ++j)
#pragma unroll
            for (int kk = 0; kk < 2; ++kk) readBfrag(buf, nsub, j, kk);
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
                for (int kk = 0; kk < 2; ++kk)
                    acc[msub * 4 + i][nsub * 2 + j] =
                        __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                            Af[msub][i][kk], Bf[nsub][j][kk],
                            acc[msub * 4 + i][nsub * 2 + j], 0, 0, 0);
        __builtin_amdgcn_s_setprio(0);
    };

    // ---- prologue: tile0 (full) + tile1.A, then q0 prefetch -------
#pragma unroll
    for (int g = 0; g < 4; ++g) stageA(0, g, 0);
#pragma unroll
    for (int g = 0; g < 4; ++g) stageB(0, g, 0);
#pragma unroll
    for (int g = 0; g < 4; ++g) stageA(1, g, 1);
    SS_VMCNT(4);
    SS_BAR();
    readA(0, 0);  // 16 tr
    readB(0, 0);  // 8 tr

    // ---- main loop: phases/staging/cert identical to gemm256.hip;
    // >15-read phases split their issue around the MFMA block.
    const int niter = ksteps / 2;
    for (int it = 0; it < niter; ++it) {
        const int t1 = 2 * it + 1;
        const int t2 = min(2 * it + 2, ksteps - 1);
        const int t3 = min(2 * it + 3, ksteps - 1);
        // P1: q0 buf0
        stageB(1, 0, t1);
        stageB(1, 1, t1);
        readB(0, 1);  // 8
        SS_LGKM(8);
        mfma16(0, 0);
        SS_BAR();
        // P2: q1
        stageB(1, 2, t1);
        stageB(1, 3, t1);
        readAhalf(0, 1, 0);  // 8
        SS_LGKM(8);
        mfma16(0, 1);
        readAhalf(0, 1, 1);  // 8, consumed P3: covered by its LGKM(0)
        SS_BAR();
        // P3: q2
        stageA(0, 0, t2);
        stageA(0, 2, t2);
        SS_LGKM(0);
        mfma16(1, 0);
        SS_BAR();
        // P4: q3; certify buf1 tile; prefetch its q0 frags
        stageA(0, 1, t2);
        stageA(0, 3, t2);
        SS_VMCNT(4);
        SS_BAR();
        readA(1, 0);  // 16
        SS_LGKM(15);  // ISA cap; waits 1 of our own, all priors done
        mfma16(1, 1);
        readB(1, 0);  // 8, consumed P5
        SS_BAR();
        // P5: q0 buf1
        stageB(0, 0, t2);
        stageB(0, 1, t2);
        readB(1, 1);
        SS_LGKM(8);
        mfma16(0, 0);
        SS_BAR();
        // P6: q1
        stageB(0, 2, t2);
        stageB(0, 3, t2);
        readAhalf(1, 1, 0);
        SS_LGKM(8);
        mfma16(0, 1);
        readAhalf(1, 1, 1);
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
        SS_LGKM(15);
        mfma16(1, 1);
        readB(0, 0);
        SS_BAR();
    }
    SS_VMCNT(0);

    // ---- epilogue: accumulate into gW -----------------------------
#pragma unroll
    for (int I = 0; I < 8; ++I) {
#pragma unroll
        for (int J = 0; J < 4; ++J) {
            const int gcol = n0 + wn * 64 + J * 16 + lrow;
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int grow = m0 + wm * 128 + I * 16 + kch * 4 + r;
                atomicAdd(&gW[(long)grow * N + gcol], acc[I][J][r]);
            }
        }
    }
}

}  // namespace

// Returns false when the shape is outside this tier.
bool ss_wgrad_tn_256(const void* dY, const void* X, void* gW, int Mo, int N,
                     int Kb, hipStream_t stream) {
    if (Mo % 256 || N % 256 || Kb % 128 || Kb < 128) return false;
    if ((long)Kb * Mo * 2 >= (1L << 31) || (long)Kb * N * 2 >= (1L << 31))
        return false;
    const int mt = Mo / 256, nt = N / 256;
    // coverage split over the batch: enough blocks for 256 CUs, each
    // split a multiple of 128 batch rows
    int split = 1;
    const int max_split = Kb / 128;
    while (split * 2 <= max_split && mt * nt * split < 256) split *= 2;
    const int kps = ((Kb / split + 127) / 128) * 128;
    const int zs = (Kb + kps - 1) / kps;
    dim3 grid(mt, nt, zs);
    dim3 blk(512);
    hipLaunchKernelGGL(wgrad_tn_256_kernel, grid, blk, 0, stream,
                       (const __bf16*)dY, (const __bf16*)X, (float*)gW, Mo,
                       N, Kb, kps);
    return true;
}
