// MFMA bf16 GEMM kernels for the MLP hot path (gfx950 / MI355X).
//
// Replaces the reference's NumPy BLAS calls (functional.py:13-21):
//   * gemm_nt:  C[M,N] = A[M,K] · B[N,K]^T (+bias) (+ReLU epilogue)
//               — forward (A=x, B=W) AND dgrad (A=dy, B=W^T copy),
//               with an optional fused ReLU-backward mask applied to A
//               while staging (mask = stashed post-ReLU output, same
//               shape as A).
//   * wgrad_tn: gW[Mo,N] += (A[Kb,Mo] ⊙ mask)^T · B[Kb,N], split-K over
//               the batch with f32 atomicAdd — the atomic IS the
//               gradient accumulation across µbatches/split-K slices
//               (reference grad accumulation: layers.py:135-136).
//   * colsum:   gb[N] += Σ_rows (dY ⊙ mask) — bias grad.
//
// Design notes (cdna_hip_programming.md):
//   * mfma_f32_16x16x32_bf16: per-wave 16×16 tile, K=32 per issue;
//     lane l holds A[l&15][(l>>4)*8 + j] / B-col[l&15], C row
//     (l>>4)*4+reg, col l&15 (verified on hardware by
//     tests/test_gpu_numerics.py::test_mfma_layout).
//   * LDS tiles padded +8 bf16 per row (16 B) → conflict-free
//     ds_read_b128 fragment reads (Guideline 4).
//   * These MLP shapes are small/memory-bound; the simple one-buffer
//     two-barrier structure is chosen for robustness at arbitrary
//     M/N/K (full bounds handling) — launch count, not MFMA peak, is
//     the budget here.

#include "common.h"

#include <cstdlib>

using bf16x4v = __attribute__((ext_vector_type(4))) __bf16;

#define BK 32
#define LDS_PAD 8
// k-group XOR swizzle for transpose-staged tiles (wgrad): spreads the
// per-instruction m-stride-8 scatter writes over banks (2-way instead
// of 16-way) while fragment reads stay 16-B-aligned ds_read_b128.
__device__ __forceinline__ int kswz(int m, int k) {
    return ((((k) >> 3) ^ (((m) >> 3) & 3)) << 3) | ((k) & 7);
}

// ---------------------------------------------------------------- gemm_nt

// Staging is split into an ISSUE half (global -> registers, with
// bounds zero-fill and the optional fused >0 mask) and a WRITE half
// (registers -> padded LDS tile) so the HBM latency of tile t+1 hides
// under tile t's MFMA work (async-STAGE split, one barrier per K-step;
// cdna_hip_programming.md Guideline 15 / T14).
template <int TILE_ROWS, int KSTEP, int NT, bool HAS_MASK>
struct StageReg {
    static constexpr int EL = TILE_ROWS * KSTEP / NT;  // 4..32
    __bf16 v[EL];

    __device__ __forceinline__ void load(const __bf16* __restrict__ src,
                                         const __bf16* __restrict__ msk,
                                         int nrows, int K, int row0, int k0,
                                         int tid) {
        const int off = tid * EL;
        const int r = off / KSTEP, c = off % KSTEP;
        const int g = row0 + r, gk = k0 + c;
        if (g < nrows && gk + EL <= K) {
#pragma unroll
            for (int ch = 0; ch < EL; ch += (EL < 8 ? 4 : 8)) {
                if constexpr (EL >= 8) {
                    bf16x8 t = *(const bf16x8*)&src[(long)g * K + gk + ch];
                    if constexpr (HAS_MASK) {
                        bf16x8 mk =
                            *(const bf16x8*)&msk[(long)g * K + gk + ch];
#pragma unroll
                        for (int i = 0; i < 8; ++i)
                            if (!(bf2f(mk[i]) > 0.f)) t[i] = (__bf16)0.f;
                    }
#pragma unroll
                    for (int i = 0; i < 8; ++i) v[ch + i] = t[i];
                } else {
                    bf16x4v t = *(const bf16x4v*)&src[(long)g * K + gk + ch];
                    if constexpr (HAS_MASK) {
                        bf16x4v mk =
                            *(const bf16x4v*)&msk[(long)g * K + gk + ch];
#pragma unroll
                        for (int i = 0; i < 4; ++i)
                            if (!(bf2f(mk[i]) > 0.f)) t[i] = (__bf16)0.f;
                    }
#pragma unroll
                    for (int i = 0; i < 4; ++i) v[ch + i] = t[i];
                }
            }
        } else {
#pragma unroll
            for (int i = 0; i < EL; ++i) {
                __bf16 t = (__bf16)0.f;
                if (g < nrows && gk + i < K) {
                    t = src[(long)g * K + gk + i];
                    if constexpr (HAS_MASK) {
                        if (!(bf2f(msk[(long)g * K + gk + i]) > 0.f))
                            t = (__bf16)0.f;
                    }
                }
                v[i] = t;
            }
        }
    }

    __device__ __forceinline__ void write(
        ushort (*__restrict__ dst)[KSTEP + LDS_PAD], int tid) const {
        const int off = tid * EL;
        const int r = off / KSTEP, c = off % KSTEP;
#pragma unroll
        for (int ch = 0; ch < EL; ch += (EL < 8 ? 4 : 8)) {
            if constexpr (EL >= 8) {
                bf16x8 t;
#pragma unroll
                for (int i = 0; i < 8; ++i) t[i] = v[ch + i];
                *(bf16x8*)&dst[r][c + ch] = t;
            } else {
                bf16x4v t;
#pragma unroll
                for (int i = 0; i < 4; ++i) t[i] = v[ch + i];
                *(bf16x4v*)&dst[r][c + ch] = t;
            }
        }
    }
};

template <int BM, int BN, int KSTEP, int WAVES_M, int WAVES_N,
          bool HAS_BIAS, bool RELU, bool HAS_MASK>
__global__ __launch_bounds__(WAVES_M * WAVES_N * 64) void gemm_nt_kernel(
    const __bf16* __restrict__ A,     // [M][K]
    const __bf16* __restrict__ B,     // [N][K]
    const __bf16* __restrict__ bias,  // [N]
    const __bf16* __restrict__ mask,  // [M][K] (A-mask source, >0 keeps)
    __bf16* __restrict__ C,           // [M][N]
    int M, int N, int K) {
    constexpr int WM = BM / WAVES_M;        // wave tile rows
    constexpr int WN = BN / WAVES_N;        // wave tile cols
    constexpr int FM = WM / 16;             // 16x16 frags per wave (M)
    constexpr int FN = WN / 16;
    constexpr int LDA = KSTEP + LDS_PAD;

    __shared__ ushort As[2][BM][LDA];
    __shared__ ushort Bs[2][BN][LDA];

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wave = tid >> 6;
    const int wm = wave / WAVES_N;
    const int wn = wave % WAVES_N;
    // XCD-aware bijective remap (T1): decode the work id n-tile-fastest
    // so the blocks sharing an A (activation) m-slice run on ONE XCD
    // and hit its L2 instead of filling all eight.
    const int gx = gridDim.x, gy = gridDim.y;
    const int nwg = gx * gy;
    const int hw = blockIdx.x + gx * blockIdx.y;
    const int xcd = hw % 8, q8 = nwg / 8, r8 = nwg % 8;
    const int wid = (xcd < r8 ? xcd * (q8 + 1)
                              : r8 * (q8 + 1) + (xcd - r8) * q8) + hw / 8;
    const int m0 = ((wid / gy) % gx) * BM;
    const int n0 = (wid % gy) * BN;

    f32x4 acc[FM][FN];
#pragma unroll
    for (int i = 0; i < FM; ++i)
#pragma unroll
        for (int j = 0; j < FN; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

    const int lrow = lane & 15;
    const int kch = lane >> 4;  // 0..3 -> k offset kch*8

    StageReg<BM, KSTEP, WAVES_M * WAVES_N * 64, HAS_MASK> ra;
    StageReg<BN, KSTEP, WAVES_M * WAVES_N * 64, false> rb;
    ra.load(A, mask, M, K, m0, 0, tid);
    rb.load(B, nullptr, N, K, n0, 0, tid);
    ra.write(As[0], tid);
    rb.write(Bs[0], tid);
    __syncthreads();

    const int nsteps = (K + KSTEP - 1) / KSTEP;
    int cur = 0;
    for (int t = 0; t < nsteps; ++t) {
        // issue next tile's global loads early (latency hides under
        // this tile's MFMA)
        if (t + 1 < nsteps) {
            ra.load(A, mask, M, K, m0, (t + 1) * KSTEP, tid);
            rb.load(B, nullptr, N, K, n0, (t + 1) * KSTEP, tid);
        }

        // ---- MFMA on the current tile ----
#pragma unroll
        for (int kk = 0; kk < KSTEP / 32; ++kk) {
            bf16x8 a_frag[FM], b_frag[FN];
#pragma unroll
            for (int i = 0; i < FM; ++i)
                a_frag[i] = *(const bf16x8*)&As[cur][wm * WM + i * 16 + lrow]
                                                   [kk * 32 + kch * 8];
#pragma unroll
            for (int j = 0; j < FN; ++j)
                b_frag[j] = *(const bf16x8*)&Bs[cur][wn * WN + j * 16 + lrow]
                                                   [kk * 32 + kch * 8];
#pragma unroll
            for (int i = 0; i < FM; ++i)
#pragma unroll
                for (int j = 0; j < FN; ++j)
                    acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        a_frag[i], b_frag[j], acc[i][j], 0, 0, 0);
        }

        if (t + 1 < nsteps) {
            ra.write(As[cur ^ 1], tid);
            rb.write(Bs[cur ^ 1], tid);
        }
        __syncthreads();
        cur ^= 1;
    }

    // ---- epilogue: bias, ReLU, bf16 store ----
#pragma unroll
    for (int i = 0; i < FM; ++i) {
#pragma unroll
        for (int j = 0; j < FN; ++j) {
            const int gcol = n0 + wn * WN + j * 16 + lrow;
            if (gcol >= N) continue;
            float bv = 0.f;
            if constexpr (HAS_BIAS) bv = bf2f(bias[gcol]);
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int grow = m0 + wm * WM + i * 16 + kch * 4 + r;
                if (grow >= M) continue;
                float v = acc[i][j][r];
                if constexpr (HAS_BIAS) v += bv;
                if constexpr (RELU) v = v > 0.f ? v : 0.f;
                C[(long)grow * N + gcol] = f2bf(v);
            }
        }
    }
}

// 256-tile 8-phase tiers (gemm256.hip / wgrad256.hip) — C++ linkage,
// declared OUTSIDE the extern "C" launcher block below
bool ss_gemm_nt_256(const void*, const void*, const void*, void*, int, int,
                    int, bool, hipStream_t);
bool ss_wgrad_tn_256(const void*, const void*, void*, int, int, int,
                     hipStream_t);

// ------------------------------------------------- gemm_nt (glds tier)

// Compute-bound tier (aligned ≥128-multiple shapes, no fused mask):
// stages tiles with gfx950 global_load_lds (direct HBM->LDS DMA, no
// VGPR round trip; the CDNA4 guide's ladder measures this as the
// dominant lever for the 128² structure).  LDS image is LANE-LINEAR
// (glds writes base+lane*16), so the bank-conflict fix is an XOR
// swizzle applied to BOTH the per-lane global SOURCE address and the
// fragment reads (rule 21): 16-B chunk c of row r lives at physical
// chunk c ^ ((r>>2)&3) — fragment-read banks become all-distinct.
template <bool HAS_BIAS, bool RELU>
__global__ __launch_bounds__(256) void gemm_nt_glds_kernel(
    const __bf16* __restrict__ A,     // [M][K], M%128==0, K%32==0
    const __bf16* __restrict__ B,     // [N][K], N%128==0
    const __bf16* __restrict__ bias,  // [N]
    __bf16* __restrict__ C,           // [M][N]
    int M, int N, int K) {
    constexpr int BM = 128, BN = 128, BKG = 32;
    typedef __attribute__((address_space(1))) const unsigned int* gptr_t;
    typedef __attribute__((address_space(3))) unsigned int* lptr_t;

    __shared__ ushort As[2][BM * BKG];  // lane-linear [128][32] bf16
    __shared__ ushort Bs[2][BN * BKG];

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wave = tid >> 6;
    const int wm = wave >> 1;  // 2x2 waves, wave tile 64x64
    const int wn = wave & 1;

    // XCD-aware bijective remap (see gemm_nt_kernel)
    const int gx = gridDim.x, gy = gridDim.y;
    const int nwg = gx * gy;
    const int hw = blockIdx.x + gx * blockIdx.y;
    const int xcd = hw % 8, q8 = nwg / 8, r8 = nwg % 8;
    const int wid = (xcd < r8 ? xcd * (q8 + 1)
                              : r8 * (q8 + 1) + (xcd - r8) * q8) + hw / 8;
    const int m0 = ((wid / gy) % gx) * BM;
    const int n0 = (wid % gy) * BN;

    f32x4 acc[4][4];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

    // staging: glds is a WAVE instruction — each of the 4 waves gets
    // its own wave-uniform LDS base (16-row span); lane l covers
    // (row = l>>2 within the span, physical 16-B chunk p = l&3).  The
    // source chunk is PRE-SWIZZLED: c = p ^ ((row>>2)&3), matching
    // the fragment-read XOR (rule 21: both sides or neither).
    const int lrow4 = lane >> 2;  // 0..15 rows within the wave span
    const int p16 = lane & 3;     // physical chunk
    auto stage = [&](int buf, int k0) {
#pragma unroll
        for (int half = 0; half < 2; ++half) {
            const int rA = half * 64 + wave * 16 + lrow4;
            const int cA = (p16 ^ ((rA >> 2) & 3)) * 8;
            __builtin_amdgcn_global_load_lds(
                (gptr_t)&A[(long)(m0 + rA) * K + k0 + cA],
                (lptr_t)&As[buf][(half * 64 + wave * 16) * BKG], 16, 0, 0);
        }
#pragma unroll
        for (int half = 0; half < 2; ++half) {
            const int rB = half * 64 + wave * 16 + lrow4;
            const int cB = (p16 ^ ((rB >> 2) & 3)) * 8;
            __builtin_amdgcn_global_load_lds(
                (gptr_t)&B[(long)(n0 + rB) * K + k0 + cB],
                (lptr_t)&Bs[buf][(half * 64 + wave * 16) * BKG], 16, 0, 0);
        }
    };

    const int lrow = lane & 15;
    const int kch = lane >> 4;

    stage(0, 0);
    __syncthreads();

    const int nsteps = K / BKG;
    int cur = 0;
    for (int t = 0; t < nsteps; ++t) {
        if (t + 1 < nsteps) stage(cur ^ 1, (t + 1) * BKG);

        bf16x8 a_frag[4], b_frag[4];
#pragma unroll
        for (int i = 0; i < 4; ++i) {
            const int row = wm * 64 + i * 16 + lrow;
            a_frag[i] = *(const bf16x8*)
                &As[cur][row * BKG + (kch ^ ((row >> 2) & 3)) * 8];
        }
#pragma unroll
        for (int j = 0; j < 4; ++j) {
            const int row = wn * 64 + j * 16 + lrow;
            b_frag[j] = *(const bf16x8*)
                &Bs[cur][row * BKG + (kch ^ ((row >> 2) & 3)) * 8];
        }
#pragma unroll
        for (int i = 0; i < 4; ++i)
#pragma unroll
            for (int j = 0; j < 4; ++j)
                acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    a_frag[i], b_frag[j], acc[i][j], 0, 0, 0);
        // __syncthreads() drains the in-flight glds (hipcc emits the
        // vmcnt(0) inside the barrier when LDS-DMA is outstanding)
        __syncthreads();
        cur ^= 1;
    }

#pragma unroll
    for (int i = 0; i < 4; ++i) {
#pragma unroll
        for (int j = 0; j < 4; ++j) {
            const int gcol = n0 + wn * 64 + j * 16 + lrow;
            float bv = 0.f;
            if constexpr (HAS_BIAS) bv = bf2f(bias[gcol]);
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int grow = m0 + wm * 64 + i * 16 + kch * 4 + r;
                float v = acc[i][j][r];
                if constexpr (HAS_BIAS) v += bv;
                if constexpr (RELU) v = v > 0.f ? v : 0.f;
                C[(long)grow * N + gcol] = f2bf(v);
            }
        }
    }
}

// ---------------------------------------------------------------- wgrad

template <int BMT, int BNT, bool HAS_MASK>
__global__ __launch_bounds__(256) void wgrad_tn_kernel(
    const __bf16* __restrict__ dY,    // [Kb][Mo]
    const __bf16* __restrict__ X,     // [Kb][N]
    const __bf16* __restrict__ mask,  // [Kb][Mo]
    float* __restrict__ gW,           // [Mo][N] (atomicAdd +=)
    float* __restrict__ gb,           // [Mo] or null: fused bias grad
    int Mo, int N, int Kb, int k_per_split,
    // optional CHUNK TABLE (deferred µbatch wgrad): int64 rows of
    // {dY*, X*, mask*}; grid.z covers chunks*split and each chunk is
    // an independent [Kb][*] pair accumulated into the same gW/gb —
    // ONE launch replaces num_µbatches launches per layer.
    const long* __restrict__ chunks, int nsplit) {
    // TN GEMM over the batch axis.  Tiles are staged in a BLOCKED
    // [BKW/4][rows/16][4][16] bf16 layout so that
    //   * staging is plain 16-B vector writes (global rows are
    //     m/n-contiguous — no software transpose), and
    //   * MFMA A/B fragments come out of LDS through the gfx950
    //     HARDWARE transpose read ds_read_b64_tr_b16
    //     (__builtin_amdgcn_ds_read_tr16_b64_v4bf16): each 16-lane
    //     group hands the crossbar one [4][16] subtile and receives
    //     its column — semantics verified on hardware by
    //     scripts/probe_tr.hip.
    // The 128x128 instantiation halves both operands' L2 re-reads for
    // wide layers (traffic ∝ dY·(N/BNT) + X·(Mo/BMT)).
    constexpr int BKW = 64;        // K-step (batch rows per stage)
    constexpr int NMA = BMT / 16;  // 16-row blocks per A tile
    constexpr int NMB = BNT / 16;
    constexpr int FM = BMT / 32;   // 16x16 frags per wave (2x2 waves)
    constexpr int FN = BNT / 32;

    __shared__ ushort At[NMA * 16 * 64];  // dY^T tile, blocked+bit-swapped
    __shared__ ushort Bt[NMB * 16 * 64];  // X^T tile, blocked+bit-swapped
    __shared__ float dbs[256 / BMT > 1 ? 256 / BMT : 2][BMT];

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wave = tid >> 6;
    const int wm = wave >> 1;  // 2x2 waves
    const int wn = wave & 1;
    // XCD-aware block remap (T1, bijective variant): the dispatcher
    // places hw block b on XCD b%8; remap so each XCD owns a
    // CONTIGUOUS run of work ids decoded n-tile-fastest — blocks that
    // share a dY m-slice then read it from ONE XCD's L2 instead of
    // filling all eight.
    const int gx = gridDim.x, gy = gridDim.y;
    const int nwg = gx * gy * gridDim.z;
    const int hw = blockIdx.x + gx * (blockIdx.y + gy * blockIdx.z);
    const int xcd = hw % 8, q = nwg / 8, r = nwg % 8;
    const int wid = (xcd < r ? xcd * (q + 1)
                             : r * (q + 1) + (xcd - r) * q) + hw / 8;
    const int bidy = wid % gy;
    const int bidx = (wid / gy) % gx;
    int bidz = wid / (gy * gx);
    if (chunks) {
        const long* row = chunks + (long)(bidz / nsplit) * 3;
        dY = (const __bf16*)row[0];
        X = (const __bf16*)row[1];
        mask = (const __bf16*)row[2];
        bidz = bidz % nsplit;
    }
    const int m0 = bidx * BMT;
    const int n0 = bidy * BNT;
    const int kbeg = bidz * k_per_split;
    const int kend = min(Kb, kbeg + k_per_split);

    f32x4 acc[FM][FN];
#pragma unroll
    for (int i = 0; i < FM; ++i)
#pragma unroll
        for (int j = 0; j < FN; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

    const bool do_db = (gb != nullptr) && (bidy == 0);
    float db_part = 0.f;

    // mblk-major image with the ksub bits 0,1 SWAPPED in the subtile
    // position (wgrad256.hip's placement): the four tr-read lane
    // groups (k-stride 2) then land on {row even|odd} x {phase 0|1}
    // of the 256-B bank rows — the optimal 2-clk pattern with NO
    // padding (the old 144-B-padded image left a 4-deep overlap on
    // banks 24-31).
    auto swsub = [](int ksub) {
        return (ksub & ~3) | ((ksub & 1) << 1) | ((ksub >> 1) & 1);
    };
    auto baddrA = [swsub](int k, int m) {
        return ((m >> 4) * 16 + swsub(k >> 2)) * 64 + (k & 3) * 16 + (m & 15);
    };
    auto baddrB = [swsub](int k, int m) {
        return ((m >> 4) * 16 + swsub(k >> 2)) * 64 + (k & 3) * 16 + (m & 15);
    };

    // staging: thread t covers ELA contiguous elems of the [BKW][BMT]
    // tile (vector fast path when fully in range)
    constexpr int ELA = BKW * BMT / 256;
    constexpr int ELB = BKW * BNT / 256;
    const int ska = (tid * ELA) / BMT, sma = (tid * ELA) % BMT;
    const int skb = (tid * ELB) / BNT, smb = (tid * ELB) % BNT;

    // glds staging (unmasked full tiles): the padded blocked image is
    // 9 aligned 16-B pieces per 144-B subtile (8 data + 1 pad), so a
    // per-lane SOURCE scatter performs the transpose while the LDS
    // dest stays lane-linear; lanes landing on the pad piece load a
    // dummy in-bounds address.  piece p -> subtile s=p/9, r=p%9;
    // r<8: k = (s/NMT)*4 + (r>>1), m = (s%NMT)*16 + (r&1)*8.
    typedef __attribute__((address_space(1))) const unsigned int* ggptr_t;
    typedef __attribute__((address_space(3))) unsigned int* glptr_t;
    // unpadded image: subtile = exactly 8 aligned 16-B pieces, so the
    // per-lane source scatter (the transpose) has NO pad lanes; piece
    // q -> position p = q/8 (mblk = p/16, ksub = swsub(p&15)), r = q%8
    // -> global (k = ksub*4 + (r>>1), m = mblk*16 + (r&1)*8).
    auto glds_stage = [&](const __bf16* __restrict__ src, ushort* img,
                          int nmt, int ld, int base_col, int gk0) {
        const int total = nmt * 16 * 8;          // pieces
        const int per_wave = total / 4;          // 4 waves
        for (int i0 = 0; i0 < per_wave; i0 += 64) {
            const int q = wave * per_wave + i0 + lane;
            const int p = q >> 3, rr = q & 7;
            const int ksub = ((p & 15) & ~3) | (((p & 15) & 1) << 1) |
                             (((p & 15) >> 1) & 1);
            const int k = ksub * 4 + (rr >> 1);
            const int m = (p >> 4) * 16 + (rr & 1) * 8;
            __builtin_amdgcn_global_load_lds(
                (ggptr_t)(src + (gk0 + k) * ld + base_col + m),
                (glptr_t)(img + (long)(wave * per_wave + i0) * 8), 16, 0,
                0);
        }
    };

    for (int k0 = kbeg; k0 < kend; k0 += BKW) {
        const bool full_tile = !HAS_MASK && m0 + BMT <= Mo &&
                               n0 + BNT <= N && k0 + BKW <= kend;
        if (full_tile) {
            glds_stage(dY, At, NMA, Mo, m0, k0);
            glds_stage(X, Bt, NMB, N, n0, k0);
            __syncthreads();  // drains the LDS-DMA (vmcnt inside)
        } else {
        {
            const int gk = k0 + ska;
            if (gk < kend && m0 + BMT <= Mo) {
#pragma unroll
                for (int ch = 0; ch < ELA; ch += 8) {
                    bf16x8 v =
                        *(const bf16x8*)&dY[(long)gk * Mo + m0 + sma + ch];
                    if constexpr (HAS_MASK) {
                        bf16x8 mk = *(const bf16x8*)&mask[(long)gk * Mo + m0 +
                                                          sma + ch];
#pragma unroll
                        for (int i = 0; i < 8; ++i)
                            if (!(bf2f(mk[i]) > 0.f)) v[i] = (__bf16)0.f;
                    }
                    *(bf16x8*)&At[baddrA(ska, sma + ch)] = v;
                }
            } else {
#pragma unroll
                for (int i = 0; i < ELA; ++i) {
                    __bf16 v = (__bf16)0.f;
                    const int gm = m0 + sma + i;
                    if (gk < kend && gm < Mo) {
                        v = dY[(long)gk * Mo + gm];
                        if constexpr (HAS_MASK) {
                            if (!(bf2f(mask[(long)gk * Mo + gm]) > 0.f))
                                v = (__bf16)0.f;
                        }
                    }
                    At[baddrA(ska, sma + i)] = *(const ushort*)&v;
                }
            }
        }
        {
            const int gk = k0 + skb;
            if (gk < kend && n0 + BNT <= N) {
#pragma unroll
                for (int ch = 0; ch < ELB; ch += 8) {
                    bf16x8 v =
                        *(const bf16x8*)&X[(long)gk * N + n0 + smb + ch];
                    *(bf16x8*)&Bt[baddrB(skb, smb + ch)] = v;
                }
            } else {
#pragma unroll
                for (int i = 0; i < ELB; ++i) {
                    __bf16 v = (__bf16)0.f;
                    const int gn = n0 + smb + i;
                    if (gk < kend && gn < N) v = X[(long)gk * N + gn];
                    Bt[baddrB(skb, smb + i)] = *(const ushort*)&v;
                }
            }
        }
        __syncthreads();
        }

        if (do_db) {
            constexpr int NKQ = 256 / BMT;       // threads stacked on k
            constexpr int KPT = BKW / NKQ;       // k values per thread
            const int m = tid % BMT, kq = tid / BMT;
#pragma unroll
            for (int j = 0; j < KPT; ++j) {
                const ushort u = At[baddrA(kq * KPT + j, m)];
                db_part += bf2f(*(const __bf16*)&u);
            }
        }

        // ---- fragments via hardware transpose read ----
        typedef __attribute__((address_space(3))) bf16x4v* lds_v4p;
        const int lcol4 = (lane & 15) * 4;
#pragma unroll
        for (int kk = 0; kk < BKW / 32; ++kk) {
            const int g2 = kk * 8 + (lane >> 4) * 2;  // k-subtile pair
            bf16x8 a_frag[FM], b_frag[FN];
#pragma unroll
            for (int i = 0; i < FM; ++i) {
                const int mblk = wm * FM + i;
                bf16x4v lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
                    (lds_v4p)&At[(mblk * 16 + swsub(g2)) * 64 + lcol4]);
                bf16x4v hi = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
                    (lds_v4p)&At[(mblk * 16 + swsub(g2 + 1)) * 64 + lcol4]);
#pragma unroll
                for (int e = 0; e < 4; ++e) {
                    a_frag[i][e] = lo[e];
                    a_frag[i][e + 4] = hi[e];
                }
            }
#pragma unroll
            for (int j = 0; j < FN; ++j) {
                const int nblk = wn * FN + j;
                bf16x4v lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
                    (lds_v4p)&Bt[(nblk * 16 + swsub(g2)) * 64 + lcol4]);
                bf16x4v hi = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
                    (lds_v4p)&Bt[(nblk * 16 + swsub(g2 + 1)) * 64 + lcol4]);
#pragma unroll
                for (int e = 0; e < 4; ++e) {
                    b_frag[j][e] = lo[e];
                    b_frag[j][e + 4] = hi[e];
                }
            }
#pragma unroll
            for (int i = 0; i < FM; ++i)
#pragma unroll
                for (int j = 0; j < FN; ++j)
                    acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        a_frag[i], b_frag[j], acc[i][j], 0, 0, 0);
        }
        __syncthreads();
    }

    const int lrow = lane & 15;
    const int kch = lane >> 4;
#pragma unroll
    for (int i = 0; i < FM; ++i) {
#pragma unroll
        for (int j = 0; j < FN; ++j) {
            const int gcol = n0 + wn * (BNT / 2) + j * 16 + lrow;
            if (gcol >= N) continue;
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int grow = m0 + wm * (BMT / 2) + i * 16 + kch * 4 + r;
                if (grow >= Mo) continue;
                atomicAdd(&gW[(long)grow * N + gcol], acc[i][j][r]);
            }
        }
    }

    if (do_db) {
        constexpr int NKQ = 256 / BMT;
        dbs[tid / BMT][tid % BMT] = db_part;
        __syncthreads();
        if (tid < BMT && m0 + tid < Mo) {
            float s4 = 0.f;
#pragma unroll
            for (int q = 0; q < NKQ; ++q) s4 += dbs[q][tid];
            atomicAdd(&gb[m0 + tid], s4);
        }
    }
}

// ---------------------------------------------------------------- colsum

// 8-wide vectorized variant: each thread owns 8 contiguous columns
// (one bf16x8 load per row), 8x the bytes in flight per thread vs the
// scalar kernel (measured 195 us at 16384x4096 = 11x off roofline —
// latency-bound scalar loads).
template <bool HAS_MASK>
__global__ __launch_bounds__(256) void colsum8_kernel(
    const __bf16* __restrict__ dY,    // [M][N]
    const __bf16* __restrict__ mask,  // [M][N]
    float* __restrict__ gb,           // [N] (atomicAdd +=)
    int M, int N, int rows_per_block) {
    const int col0 = (blockIdx.x * 256 + threadIdx.x) * 8;
    const int r0 = blockIdx.y * rows_per_block;
    const int r1 = min(M, r0 + rows_per_block);
    if (col0 >= N) return;
    float s[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
    // 4-row unroll: four independent 16-B loads in flight per thread
    // per iteration (the single-row loop was latency-serialized —
    // measured 78 µs at 16384×4096 vs the ~17 µs traffic floor)
    int r = r0;
    for (; r + 4 <= r1; r += 4) {
        bf16x8 v0 = *(const bf16x8*)(dY + (long)(r + 0) * N + col0);
        bf16x8 v1 = *(const bf16x8*)(dY + (long)(r + 1) * N + col0);
        bf16x8 v2 = *(const bf16x8*)(dY + (long)(r + 2) * N + col0);
        bf16x8 v3 = *(const bf16x8*)(dY + (long)(r + 3) * N + col0);
        if constexpr (HAS_MASK) {
            bf16x8 m0 = *(const bf16x8*)(mask + (long)(r + 0) * N + col0);
            bf16x8 m1 = *(const bf16x8*)(mask + (long)(r + 1) * N + col0);
            bf16x8 m2 = *(const bf16x8*)(mask + (long)(r + 2) * N + col0);
            bf16x8 m3 = *(const bf16x8*)(mask + (long)(r + 3) * N + col0);
#pragma unroll
            for (int e = 0; e < 8; ++e) {
                if (bf2f(m0[e]) > 0.f) s[e] += bf2f(v0[e]);
                if (bf2f(m1[e]) > 0.f) s[e] += bf2f(v1[e]);
                if (bf2f(m2[e]) > 0.f) s[e] += bf2f(v2[e]);
                if (bf2f(m3[e]) > 0.f) s[e] += bf2f(v3[e]);
            }
        } else {
#pragma unroll
            for (int e = 0; e < 8; ++e)
                s[e] += bf2f(v0[e]) + bf2f(v1[e]) + bf2f(v2[e]) +
                        bf2f(v3[e]);
        }
    }
    for (; r < r1; ++r) {
        bf16x8 v = *(const bf16x8*)(dY + (long)r * N + col0);
        if constexpr (HAS_MASK) {
            bf16x8 m = *(const bf16x8*)(mask + (long)r * N + col0);
#pragma unroll
            for (int e = 0; e < 8; ++e)
                if (bf2f(m[e]) > 0.f) s[e] += bf2f(v[e]);
        } else {
#pragma unroll
            for (int e = 0; e < 8; ++e) s[e] += bf2f(v[e]);
        }
    }
#pragma unroll
    for (int e = 0; e < 8; ++e) atomicAdd(&gb[col0 + e], s[e]);
}

template <bool HAS_MASK>
__global__ __launch_bounds__(256) void colsum_kernel(
    const __bf16* __restrict__ dY,    // [M][N]
    const __bf16* __restrict__ mask,  // [M][N]
    float* __restrict__ gb,           // [N] (atomicAdd +=)
    int M, int N, int rows_per_block) {
    const int col = blockIdx.x * 256 + threadIdx.x;
    const int r0 = blockIdx.y * rows_per_block;
    const int r1 = min(M, r0 + rows_per_block);
    if (col >= N) return;
    float s = 0.f;
    for (int r = r0; r < r1; ++r) {
        float v = bf2f(dY[(long)r * N + col]);
        if constexpr (HAS_MASK) {
            if (!(bf2f(mask[(long)r * N + col]) > 0.f)) v = 0.f;
        }
        s += v;
    }
    atomicAdd(&gb[col], s);
}

// ---------------------------------------------------------------- launchers

extern "C" {

void ss_gemm_nt(const void* A, const void* B, const void* bias,
                const void* mask, void* C, int M, int N, int K,
                bool relu, hipStream_t stream) {
    const bool has_bias = bias != nullptr;
    const bool has_mask = mask != nullptr;

    auto launch = [&](auto bm_tag, auto bias_tag, auto relu_tag, auto mask_tag) {
        constexpr int BM = decltype(bm_tag)::value;
        constexpr bool HB = decltype(bias_tag)::value;
        constexpr bool RL = decltype(relu_tag)::value;
        constexpr bool HM = decltype(mask_tag)::value;
        constexpr int BN = BM == 128 ? 128 : 64;
        constexpr int WAVES_M = BM == 32 ? 1 : 2;
        // 4 waves 2x2 everywhere ≥64: 8-wave 128² measured WORSE
        // (623 vs 720 TF @8k³) without the full 8-phase schedule —
        // exactly the guide's T3/T5 regime gate
        constexpr int WAVES_N = BM == 32 ? 4 : 2;
        dim3 block(WAVES_M * WAVES_N * 64);
        dim3 grid(cdiv(M, BM), cdiv(N, BN));
        // Measured: KSTEP=64 is throughput-neutral at these shapes
        // (barrier savings offset by the occupancy drop 7->4
        // blocks/CU), so the deep K-step stays off; the template
        // keeps it one constant away for wider models.
        if (false && K >= 512 && BM >= 64) {
            hipLaunchKernelGGL(
                (gemm_nt_kernel<BM, BN, 64, WAVES_M, WAVES_N, HB, RL, HM>),
                grid, block, 0, stream,
                (const __bf16*)A, (const __bf16*)B, (const __bf16*)bias,
                (const __bf16*)mask, (__bf16*)C, M, N, K);
        } else {
            hipLaunchKernelGGL(
                (gemm_nt_kernel<BM, BN, 32, WAVES_M, WAVES_N, HB, RL, HM>),
                grid, block, 0, stream,
                (const __bf16*)A, (const __bf16*)B, (const __bf16*)bias,
                (const __bf16*)mask, (__bf16*)C, M, N, K);
        }
    };

    using T = std::true_type;
    using F = std::false_type;
    using B128 = std::integral_constant<int, 128>;
    using B64 = std::integral_constant<int, 64>;
    using B32 = std::integral_constant<int, 32>;

#define DISPATCH(BMT)                                                  \
    if (has_bias && relu)        launch(BMT{}, T{}, T{}, F{});         \
    else if (has_bias)           launch(BMT{}, T{}, F{}, F{});         \
    else if (has_mask)           launch(BMT{}, F{}, F{}, T{});         \
    else if (relu)               launch(BMT{}, F{}, T{}, F{});         \
    else                         launch(BMT{}, F{}, F{}, F{});

    // widest tier first: 256-tile 8-phase kernel (gemm256.hip) for
    // aligned wide shapes — measured 1.3-1.6x the 128-glds tier
    // (1027-1293 TF vs 692-811 at 4096-16384 x 4096^2, zero LDS bank
    // conflicts; scripts/test_gemm256.py).  SS_GEMM256=0 disables.
    if (!has_mask && N >= 512 && (long)(M / 256) * (N / 256) >= 128) {
        static int en256 = -1;
        if (en256 < 0) {
            const char* e = getenv("SS_GEMM256");
            en256 = e ? atoi(e) : 1;
        }
        if (en256 && ss_gemm_nt_256(A, B, bias, C, M, N, K, relu, stream))
            return;
    }
    // compute-bound tier: aligned shapes with no fused mask go to the
    // glds-staged kernel (direct HBM->LDS DMA)
    if (!has_mask && N >= 128 && M % 128 == 0 && N % 128 == 0 &&
        K % 32 == 0 && (long)(M / 128) * (N / 128) >= 512) {
        dim3 grid(M / 128, N / 128);
        dim3 blk(256);
        if (has_bias && relu)
            hipLaunchKernelGGL((gemm_nt_glds_kernel<true, true>), grid, blk,
                               0, stream, (const __bf16*)A, (const __bf16*)B,
                               (const __bf16*)bias, (__bf16*)C, M, N, K);
        else if (has_bias)
            hipLaunchKernelGGL((gemm_nt_glds_kernel<true, false>), grid, blk,
                               0, stream, (const __bf16*)A, (const __bf16*)B,
                               (const __bf16*)bias, (__bf16*)C, M, N, K);
        else if (relu)
            hipLaunchKernelGGL((gemm_nt_glds_kernel<false, true>), grid, blk,
                               0, stream, (const __bf16*)A, (const __bf16*)B,
                               (const __bf16*)bias, (__bf16*)C, M, N, K);
        else
            hipLaunchKernelGGL((gemm_nt_glds_kernel<false, false>), grid, blk,
                               0, stream, (const __bf16*)A, (const __bf16*)B,
                               (const __bf16*)bias, (__bf16*)C, M, N, K);
        return;
    }
    // 128x128 only when the grid still covers the CUs with >=2
    // blocks each (1 block/CU = 4 waves starves latency hiding)
    if (N >= 128 && (long)cdiv(M, 128) * cdiv(N, 128) >= 512) {
        DISPATCH(B128)
    } else if (M < 48 ||
               (long)cdiv(M, 64) * cdiv(N, 64) < 192) {
        // small-M / small-grid shapes: the 32-row config doubles the
        // block count (latency-bound regime — µbatched schedules)
        DISPATCH(B32)
    } else {
        DISPATCH(B64)
    }
#undef DISPATCH
}

void ss_colsum(const void*, const void*, void*, int, int, hipStream_t);

void ss_wgrad_tn(const void* dY, const void* X, const void* mask, void* gW,
                 void* gb, int Mo, int N, int Kb, int split_k,
                 hipStream_t stream) {
    // widest tier first: 8-phase 256-tile TN kernel for aligned wide
    // unmasked shapes (bias grad via the standalone colsum kernel).
    // SS_WGRAD256=0 disables.
    // >= 64 output tiles: below that the split-K short loops lose to
    // the 128-tile kernel (measured 440 vs 514 TF at 1024^2 x 16384)
    if (!mask && split_k <= 0 && Mo >= 512 && N >= 512 &&
        (long)(Mo / 256) * (N / 256) >= 64) {
        static int en = -1;
        if (en < 0) {
            const char* e = getenv("SS_WGRAD256");
            en = e ? atoi(e) : 1;
        }
        if (en && ss_wgrad_tn_256(dY, X, gW, Mo, N, Kb, stream)) {
            if (gb) ss_colsum(dY, nullptr, gb, Kb, Mo, stream);
            return;
        }
    }
    // big tiles only pay when both dims are wide enough that the
    // grid still covers the CUs AND operand re-reads dominate
    // (measured: at 256-wide layers the 64-config's block count wins)
    bool big = (Mo >= 256 && N >= 512) || (Mo >= 512 && N >= 256);
    // debug/tuning overrides (read once)
    static int env_big = []() {
        const char* e = getenv("SS_WGRAD_BIG");
        return e ? atoi(e) : -1;
    }();
    if (env_big >= 0) big = env_big != 0;
    // experimental 128x256 tile (halves dY re-reads when N spans many
    // 128-tiles but Mo is narrow — the flagship first layer's
    // 16384x256x784 masked wgrad): SS_WGRAD_N256=1 enables
    static int env_n256 = []() {
        const char* e = getenv("SS_WGRAD_N256");
        return e ? atoi(e) : 0;
    }();
    const bool n256 = env_n256 && Mo >= 128 && N >= 640;
    const int bm = n256 ? 128 : (big ? 128 : 64);
    const int bn = n256 ? 256 : (big ? 128 : 64);
    // pick split for ~512 blocks total (2 per CU): more split
    // duplicates the atomic-epilogue traffic (measured: 1024² best at
    // exactly the coverage split; over-splitting by K depth cost
    // +50%), less starves the CUs
    if (split_k <= 0) {
        const int tiles = cdiv(Mo, bm) * cdiv(N, bn);
        split_k = cdiv(512, tiles);
        // round DOWN to a power of two (even k-slice boundaries;
        // measured best at 256×784)
        while (split_k & (split_k - 1)) split_k &= split_k - 1;
        const int max_split = cdiv(Kb, 64);
        if (split_k > max_split) split_k = max_split;
    }
    int k_per_split = cdiv(cdiv(Kb, split_k), 64) * 64;  // BKW=64
    split_k = cdiv(Kb, k_per_split);
    dim3 grid(cdiv(Mo, bm), cdiv(N, bn), split_k);
    dim3 block(256);
#define WLAUNCH(BMT, BNT, HM)                                               \
    hipLaunchKernelGGL((wgrad_tn_kernel<BMT, BNT, HM>), grid, block, 0,     \
                       stream, (const __bf16*)dY, (const __bf16*)X,         \
                       (const __bf16*)mask, (float*)gW, (float*)gb, Mo, N,  \
                       Kb, k_per_split, (const long*)nullptr, 1)
    if (n256) {
        if (mask) WLAUNCH(128, 256, true); else WLAUNCH(128, 256, false);
    } else if (big) {
        if (mask) WLAUNCH(128, 128, true); else WLAUNCH(128, 128, false);
    } else {
        if (mask) WLAUNCH(64, 64, true); else WLAUNCH(64, 64, false);
    }
#undef WLAUNCH
}

void ss_wgrad_tn_multi(const void* chunk_table, int nchunks, bool has_mask,
                       void* gW, void* gb, int Mo, int N, int Kb_chunk,
                       int split_k, hipStream_t stream) {
    const bool big = (Mo >= 256 && N >= 512) || (Mo >= 512 && N >= 256);
    const int bm = big ? 128 : 64, bn = big ? 128 : 64;
    if (split_k <= 0) {
        const int tiles = cdiv(Mo, bm) * cdiv(N, bn) * nchunks;
        split_k = cdiv(512, tiles);
        while (split_k & (split_k - 1)) split_k &= split_k - 1;
        const int max_split = cdiv(Kb_chunk, 64);
        if (split_k > max_split) split_k = max_split;
    }
    int k_per_split = cdiv(cdiv(Kb_chunk, split_k), 64) * 64;
    split_k = cdiv(Kb_chunk, k_per_split);
    dim3 grid(cdiv(Mo, bm), cdiv(N, bn), split_k * nchunks);
    dim3 block(256);
#define WLAUNCHM(BMT, BNT, HM)                                              \
    hipLaunchKernelGGL((wgrad_tn_kernel<BMT, BNT, HM>), grid, block, 0,     \
                       stream, (const __bf16*)nullptr,                      \
                       (const __bf16*)nullptr, (const __bf16*)nullptr,      \
                       (float*)gW, (float*)gb, Mo, N, Kb_chunk,             \
                       k_per_split, (const long*)chunk_table, split_k)
    if (big) {
        if (has_mask) WLAUNCHM(128, 128, true); else WLAUNCHM(128, 128, false);
    } else {
        if (has_mask) WLAUNCHM(64, 64, true); else WLAUNCHM(64, 64, false);
    }
#undef WLAUNCHM
}

void ss_colsum(const void* dY, const void* mask, void* gb, int M, int N,
               hipStream_t stream) {
    // enough blocks to cover the CUs (256 CUs; Guideline 11)
    const bool vec8 = (N % 8 == 0) && N >= 2048;
    const int cols_per_block = vec8 ? 2048 : 256;
    int col_tiles = cdiv(N, cols_per_block);
    int splits = cdiv(512, col_tiles);
    if (splits > cdiv(M, 64)) splits = cdiv(M, 64);
    if (splits < 1) splits = 1;
    int rows_per_block = cdiv(M, splits);
    dim3 grid(col_tiles, splits);
    dim3 block(256);
    if (vec8) {
        if (mask)
            hipLaunchKernelGGL((colsum8_kernel<true>), grid, block, 0,
                               stream, (const __bf16*)dY,
                               (const __bf16*)mask, (float*)gb, M, N,
                               rows_per_block);
        else
            hipLaunchKernelGGL((colsum8_kernel<false>), grid, block, 0,
                               stream, (const __bf16*)dY,
                               (const __bf16*)mask, (float*)gb, M, N,
                               rows_per_block);
        return;
    }
    if (mask)
        hipLaunchKernelGGL((colsum_kernel<true>), grid, block, 0, stream,
                           (const __bf16*)dY, (const __bf16*)mask, (float*)gb,
                           M, N, rows_per_block);
    else
        hipLaunchKernelGGL((colsum_kernel<false>), grid, block, 0, stream,
                           (const __bf16*)dY, (const __bf16*)mask, (float*)gb,
                           M, N, rows_per_block);
}

}  // extern "C"
