// Persistent fused MLP forward (serving path, gfx950 / MI355X).
//
// One kernel runs the ENTIRE layer chain: each block owns 64 rows of
// the batch; activations stay RESIDENT in LDS between layers (never
// round-tripping through HBM), weights stream K-chunked through a
// double-buffered LDS tile (L2-resident: every block reads the same
// ~131 KB/layer), and the head emits per-row argmax directly.  This
// removes the per-kernel launch/ramp floor that dominates the
// forward-only serving loop (eager: ~59 µs of GEMM kernels + gaps per
// 16384-batch; measured graphs do NOT reclaim the gaps — replay floor
// + input copy lose, profiles/r02_evidence.md).
//
// Shape contract (checked by the launcher, fallback = eager path):
//   M % 64 == 0; hidden width H == 256 (template); 1..8 hidden
//   layers (ReLU); head width <= 16; argmax output (softmax is
//   monotone — serving prediction needs no exp).
//
// Reference scope: the full forward of compute_accuracy / serving
// (reference train.py:21-47) as one fused op.

#include "common.h"

namespace {

#define FPAD 8  // LDS row padding (ushorts) against bank conflicts

// 64×KSTEP A-chunk (or 256×KSTEP B-chunk) register-staged load, with
// zero fill beyond (nrows, K) — same structure as gemm.hip's StageReg.
template <int TILE_ROWS, int KSTEP, int NT>
struct FStage {
    static constexpr int EL = TILE_ROWS * KSTEP / NT;
    __bf16 v[EL];

    __device__ __forceinline__ void load(const __bf16* __restrict__ src,
                                         int nrows, int K, int row0, int k0,
                                         int tid) {
        const int off = tid * EL;
        const int r = off / KSTEP, c = off % KSTEP;
        const int g = row0 + r, gk = k0 + c;
        if (g < nrows && gk + EL <= K) {
#pragma unroll
            for (int ch = 0; ch < EL; ch += 8) {
                bf16x8 t = *(const bf16x8*)&src[(long)g * K + gk + ch];
#pragma unroll
                for (int i = 0; i < 8; ++i) v[ch + i] = t[i];
            }
        } else {
#pragma unroll
            for (int i = 0; i < EL; ++i)
                v[i] = (g < nrows && gk + i < K)
                           ? src[(long)g * K + gk + i]
                           : (__bf16)0.f;
        }
    }

    __device__ __forceinline__ void write(
        ushort (*__restrict__ dst)[KSTEP + FPAD], int tid) const {
        const int off = tid * EL;
        const int r = off / KSTEP, c = off % KSTEP;
#pragma unroll
        for (int ch = 0; ch < EL; ch += 8)
            *(bf16x8*)&dst[r][c + ch] = *(const bf16x8*)&v[ch];
    }
};

// desc rows (int64): {W_ptr, bias_ptr} for layer l; layer widths are
// in_dim→H (ReLU), H→H (ReLU) × (nhidden-1), H→cout (head, no act).
template <int H>
__global__ __launch_bounds__(256, 1) void fused_mlp_fwd_kernel(
    const __bf16* __restrict__ X,  // [M][in_dim]
    const long* __restrict__ desc,
    int M, int in_dim, int nhidden, int cout,
    int* __restrict__ out) {  // [M] argmax
    constexpr int KS = 64;
    __shared__ ushort act[2][64][H + FPAD];
    __shared__ ushort ws[2][H][KS + FPAD];

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wave = tid >> 6;  // 0..3; wave owns cols [wave*64, +64)
    const int lrow = lane & 15;
    const int kch = lane >> 4;
    const int m0 = blockIdx.x * 64;

    f32x4 acc[4][4];
    // layer-0 A-chunk staging aliases the (then-unused) act[1] buffer:
    // two [64][KS+FPAD] chunk tiles fit in one act buffer with room to
    // spare (2*64*(KS+FPAD) = 9216 ushorts vs 64*(H+FPAD) = 16896)
    auto axsp = (ushort(*)[KS + FPAD]) & act[1][0][0];
    // chunk buffer b occupies rows [b*64, b*64+64) of axsp

    FStage<64, KS, 256> ra;
    FStage<H, KS, 256> rb;

    int cur_act = 0;  // which act buffer holds the CURRENT layer input
    for (int l = 0; l < nhidden + 1; ++l) {
        const __bf16* W = (const __bf16*)desc[l * 2];
        const __bf16* bias = (const __bf16*)desc[l * 2 + 1];
        const int K = (l == 0) ? in_dim : H;
        const int nsteps = (K + KS - 1) / KS;
        const int brows = (l == nhidden) ? cout : H;

#pragma unroll
        for (int i = 0; i < 4; ++i)
#pragma unroll
            for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

        // prologue: stage chunk 0
        if (l == 0) {
            ra.load(X, M, K, m0, 0, tid);
            ra.write(axsp, tid);
        }
        rb.load(W, brows, K, 0, 0, tid);
        rb.write(ws[0], tid);
        __syncthreads();

        int cur = 0;
        for (int t = 0; t < nsteps; ++t) {
            if (t + 1 < nsteps) {
                if (l == 0) ra.load(X, M, K, m0, (t + 1) * KS, tid);
                rb.load(W, brows, K, 0, (t + 1) * KS, tid);
            }
#pragma unroll
            for (int kk = 0; kk < KS / 32; ++kk) {
                bf16x8 a_frag[4], b_frag[4];
#pragma unroll
                for (int i = 0; i < 4; ++i) {
                    if (l == 0)
                        a_frag[i] = *(const bf16x8*)&axsp
                            [cur * 64 + i * 16 + lrow][kk * 32 + kch * 8];
                    else
                        a_frag[i] = *(const bf16x8*)&act[cur_act]
                            [i * 16 + lrow][t * KS + kk * 32 + kch * 8];
                }
#pragma unroll
                for (int j = 0; j < 4; ++j)
                    b_frag[j] = *(const bf16x8*)&ws[cur]
                        [wave * 64 + j * 16 + lrow][kk * 32 + kch * 8];
#pragma unroll
                for (int i = 0; i < 4; ++i)
#pragma unroll
                    for (int j = 0; j < 4; ++j)
                        acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                            a_frag[i], b_frag[j], acc[i][j], 0, 0, 0);
            }
            if (t + 1 < nsteps) {
                if (l == 0) ra.write(axsp + (cur ^ 1) * 64, tid);
                rb.write(ws[cur ^ 1], tid);
            }
            __syncthreads();
            cur ^= 1;
        }

        if (l < nhidden) {
            // epilogue: bias + ReLU → the OTHER act buffer (bf16).
            // Layer 0 writes act[0] (its staging aliased act[1]);
            // later layers ping-pong.
            const int dst = (l == 0) ? 0 : (cur_act ^ 1);
#pragma unroll
            for (int j = 0; j < 4; ++j) {
                const int col = wave * 64 + j * 16 + lrow;
                const float bv = bias ? bf2f(bias[col]) : 0.f;
#pragma unroll
                for (int i = 0; i < 4; ++i)
#pragma unroll
                    for (int r = 0; r < 4; ++r) {
                        const int row = i * 16 + kch * 4 + r;
                        const float v = acc[i][j][r] + bv;
                        *(__bf16*)&act[dst][row][col] =
                            f2bf(v > 0.f ? v : 0.f);
                    }
            }
            __syncthreads();
            cur_act = dst;
        } else {
            // head: logits → per-row argmax.  Wave w computed cols
            // [w*64,+64) but cout<=16 means only wave 0's j==0 frag
            // holds real columns; every wave writes its frag so the
            // store pattern is uniform, then 64 threads reduce.
            __shared__ float logits[64][16 + 1];
            if (wave == 0) {
#pragma unroll
                for (int i = 0; i < 4; ++i)
#pragma unroll
                    for (int r = 0; r < 4; ++r) {
                        const int row = i * 16 + kch * 4 + r;
                        const int col = lrow;
                        float v = acc[i][0][r];
                        if (bias && col < cout) v += bf2f(bias[col]);
                        logits[row][col] = v;
                    }
            }
            __syncthreads();
            if (tid < 64) {
                const int grow = m0 + tid;
                if (grow < M) {
                    float best = logits[tid][0];
                    int bi = 0;
                    for (int c = 1; c < cout; ++c) {
                        const float v = logits[tid][c];
                        if (v > best) { best = v; bi = c; }
                    }
                    out[grow] = bi;
                }
            }
        }
    }
}

}  // namespace

// Returns false when the shape is outside the fused tier.
bool ss_fused_mlp_fwd(const void* x, const void* desc, int M, int in_dim,
                      int nhidden, int cout, void* out,
                      hipStream_t stream) {
    if (M % 64 || nhidden < 1 || nhidden > 8 || cout > 16 || cout < 1)
        return false;
    hipLaunchKernelGGL((fused_mlp_fwd_kernel<256>), dim3(M / 64), dim3(256),
                       0, stream, (const __bf16*)x, (const long*)desc, M,
                       in_dim, nhidden, cout, (int*)out);
    return true;
}
