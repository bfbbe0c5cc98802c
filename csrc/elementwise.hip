// Elementwise / rowwise / optimizer kernels (gfx950, MI355X).
//
// Replaces the reference's NumPy ops:
//   * relu fwd/bwd            (functional.py:4-10; hot path uses the
//                              fused GEMM epilogue/mask instead)
//   * row softmax fwd/bwd     (functional.py:24-35 — with CORRECT
//                              per-row max, unlike the reference's
//                              global-max quirk at functional.py:26)
//   * fused loss heads        (softmax∘MSE per functional.py:43-44 +
//                              softmax jacobian; softmax-cross-entropy)
//   * multi-tensor SGD        (optimizer.py:10-13) — ONE launch updates
//                              every stage parameter: f32 master -=
//                              lr·f32 grad, re-emitting the bf16 copy
//                              AND the transposed bf16 copy (which
//                              makes dgrad an NT GEMM).
//
// All memory-bound: bf16 traffic vectorized as ushort8 (16 B/lane,
// Guideline 13), f32 math internally, wave-shuffle row reductions.

#include "common.h"

// ------------------------------------------------------------- relu

__global__ __launch_bounds__(256) void relu_fwd_kernel(
    const __bf16* __restrict__ x, __bf16* __restrict__ y, long n) {
    long i = (long)(blockIdx.x * 256 + threadIdx.x) * 8;
    const long stride = (long)gridDim.x * 256 * 8;
    for (; i + 8 <= n; i += stride) {
        bf16x8 v = *(const bf16x8*)(x + i);
        bf16x8 o;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            float f = bf2f(v[j]);
            o[j] = f2bf(f > 0.f ? f : 0.f);
        }
        *(bf16x8*)(y + i) = o;
    }
    if (i < n) {
        for (; i < n; ++i) {
            float f = bf2f(x[i]);
            y[i] = f2bf(f > 0.f ? f : 0.f);
        }
    }
}

__global__ __launch_bounds__(256) void relu_bwd_kernel(
    const __bf16* __restrict__ dy, const __bf16* __restrict__ y,
    __bf16* __restrict__ dx, long n) {
    long i = (long)(blockIdx.x * 256 + threadIdx.x) * 8;
    const long stride = (long)gridDim.x * 256 * 8;
    for (; i + 8 <= n; i += stride) {
        bf16x8 g = *(const bf16x8*)(dy + i);
        bf16x8 m = *(const bf16x8*)(y + i);
        bf16x8 o;
#pragma unroll
        for (int j = 0; j < 8; ++j)
            o[j] = bf2f(m[j]) > 0.f ? g[j] : (__bf16)0.f;
        *(bf16x8*)(dx + i) = o;
    }
    if (i < n) {
        for (; i < n; ++i)
            dx[i] = bf2f(y[i]) > 0.f ? dy[i] : (__bf16)0.f;
    }
}

// ------------------------------------------------------- row reductions

__device__ __forceinline__ float wave_max(float v) {
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
        v = fmaxf(v, __shfl_xor(v, off, 64));
    return v;
}

__device__ __forceinline__ float wave_sum(float v) {
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
        v += __shfl_xor(v, off, 64);
    return v;
}

// one wave per row; 4 rows per block
__global__ __launch_bounds__(256) void softmax_fwd_kernel(
    const __bf16* __restrict__ x, __bf16* __restrict__ s, int B, int C) {
    const int lane = threadIdx.x & 63;
    const int row = blockIdx.x * 4 + (threadIdx.x >> 6);
    if (row >= B) return;
    const __bf16* xr = x + (long)row * C;
    float m = -1e30f;
    for (int c = lane; c < C; c += 64) m = fmaxf(m, bf2f(xr[c]));
    m = wave_max(m);
    float sum = 0.f;
    for (int c = lane; c < C; c += 64) sum += __expf(bf2f(xr[c]) - m);
    sum = wave_sum(sum);
    const float inv = 1.f / sum;
    __bf16* sr = s + (long)row * C;
    for (int c = lane; c < C; c += 64)
        sr[c] = f2bf(__expf(bf2f(xr[c]) - m) * inv);
}

// dx = s * (dy - rowsum(s*dy)), from stashed OUTPUT s
__global__ __launch_bounds__(256) void softmax_bwd_kernel(
    const __bf16* __restrict__ dy, const __bf16* __restrict__ s,
    __bf16* __restrict__ dx, int B, int C) {
    const int lane = threadIdx.x & 63;
    const int row = blockIdx.x * 4 + (threadIdx.x >> 6);
    if (row >= B) return;
    const __bf16* dyr = dy + (long)row * C;
    const __bf16* sr = s + (long)row * C;
    float dot = 0.f;
    for (int c = lane; c < C; c += 64) dot += bf2f(sr[c]) * bf2f(dyr[c]);
    dot = wave_sum(dot);
    __bf16* dxr = dx + (long)row * C;
    for (int c = lane; c < C; c += 64)
        dxr[c] = f2bf(bf2f(sr[c]) * (bf2f(dyr[c]) - dot));
}

// fused softmax∘MSE head backward: g = -2(t-s)/GB;
// dz = s * (g - rowsum(s*g))   (reference functional.py:30-35,43-44)
__global__ __launch_bounds__(256) void head_mse_bwd_kernel(
    const __bf16* __restrict__ s, const __bf16* __restrict__ t,
    __bf16* __restrict__ dz, int B, int C, float inv_gb) {
    const int lane = threadIdx.x & 63;
    const int row = blockIdx.x * 4 + (threadIdx.x >> 6);
    if (row >= B) return;
    const __bf16* sr = s + (long)row * C;
    const __bf16* tr = t + (long)row * C;
    float dot = 0.f;
    for (int c = lane; c < C; c += 64) {
        const float sv = bf2f(sr[c]);
        const float g = -2.f * (bf2f(tr[c]) - sv) * inv_gb;
        dot += sv * g;
    }
    dot = wave_sum(dot);
    __bf16* dzr = dz + (long)row * C;
    for (int c = lane; c < C; c += 64) {
        const float sv = bf2f(sr[c]);
        const float g = -2.f * (bf2f(tr[c]) - sv) * inv_gb;
        dzr[c] = f2bf(sv * (g - dot));
    }
}

// fused softmax-cross-entropy backward: dz = (s - t)/GB
__global__ __launch_bounds__(256) void head_xent_bwd_kernel(
    const __bf16* __restrict__ s, const __bf16* __restrict__ t,
    __bf16* __restrict__ dz, long n, float inv_gb) {
    long i = (long)(blockIdx.x * 256 + threadIdx.x) * 8;
    const long stride = (long)gridDim.x * 256 * 8;
    for (; i + 8 <= n; i += stride) {
        bf16x8 sv = *(const bf16x8*)(s + i);
        bf16x8 tv = *(const bf16x8*)(t + i);
        bf16x8 o;
#pragma unroll
        for (int j = 0; j < 8; ++j)
            o[j] = f2bf((bf2f(sv[j]) - bf2f(tv[j])) * inv_gb);
        *(bf16x8*)(dz + i) = o;
    }
    if (i < n)
        for (; i < n; ++i)
            dz[i] = f2bf((bf2f(s[i]) - bf2f(t[i])) * inv_gb);
}

// ------------------------------------------------------- row argmax

// eval/serving helper: one wave per row (like softmax); returns the
// index of the row maximum as int32.  torch's ROCm argmax on skinny
// (B x 10) bf16 tensors measured ~1.3 ms at B=16384 — this kernel is
// the same wave-reduction pattern as softmax_fwd (~5 µs).
__global__ __launch_bounds__(256) void row_argmax_kernel(
    const __bf16* __restrict__ x, int* __restrict__ out, int B, int C) {
    const int lane = threadIdx.x & 63;
    const int row = blockIdx.x * 4 + (threadIdx.x >> 6);
    if (row >= B) return;
    const __bf16* xr = x + (long)row * C;
    float best = -1e30f;
    int bi = 0;
    for (int c = lane; c < C; c += 64) {
        const float v = bf2f(xr[c]);
        if (v > best) {
            best = v;
            bi = c;
        }
    }
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
        const float ov = __shfl_xor(best, off, 64);
        const int oi = __shfl_xor(bi, off, 64);
        if (ov > best || (ov == best && oi < bi)) {
            best = ov;
            bi = oi;
        }
    }
    if (lane == 0) out[row] = bi;
}

// --------------------------------------------------------- fused SGD

// desc row (int64 x 8): master*, grad*, lp*, lpt*, numel, cols, start,
// velocity* (0 when momentum is off).  One launch updates every stage
// parameter: master -= lr * (grad + wd*master [+ momentum]), and
// re-emits the bf16 compute copy and the transposed bf16 copy.
__global__ __launch_bounds__(256) void sgd_multi_kernel(
    const long* __restrict__ desc, int ntensors, long total, float lr,
    float momentum, float weight_decay) {
    long e = (long)blockIdx.x * 256 + threadIdx.x;
    const long stride = (long)gridDim.x * 256;
    for (; e < total; e += stride) {
        // binary search the tensor whose [start, start+numel) holds e
        int lo = 0, hi = ntensors - 1;
        while (lo < hi) {
            const int mid = (lo + hi + 1) >> 1;
            if (desc[mid * 8 + 6] <= e) lo = mid;
            else hi = mid - 1;
        }
        const long* d = desc + lo * 8;
        const long i = e - d[6];
        float* master = (float*)d[0];
        const float* grad = (const float*)d[1];
        float g = grad[i];
        if (weight_decay != 0.f) g += weight_decay * master[i];
        if (d[7]) {
            float* vel = (float*)d[7];
            g = momentum * vel[i] + g;
            vel[i] = g;
        }
        const float m = master[i] - lr * g;
        master[i] = m;
        const __bf16 v = f2bf(m);
        if (d[2]) ((__bf16*)d[2])[i] = v;
        if (d[3]) {
            const long cols = d[5];
            const long rows = d[4] / cols;
            ((__bf16*)d[3])[(i % cols) * rows + i / cols] = v;
        }
    }
}

// ---------------------------------------------------------------- launchers

extern "C" {

static inline int ew_grid(long n) {
    long blocks = (n / 8 + 255) / 256;
    if (blocks < 1) blocks = 1;
    if (blocks > 2048) blocks = 2048;  // grid-stride the rest (G11)
    return (int)blocks;
}

void ss_relu_fwd(const void* x, void* y, long n, hipStream_t st) {
    hipLaunchKernelGGL(relu_fwd_kernel, dim3(ew_grid(n)), dim3(256), 0, st,
                       (const __bf16*)x, (__bf16*)y, n);
}

void ss_relu_bwd(const void* dy, const void* y, void* dx, long n,
                 hipStream_t st) {
    hipLaunchKernelGGL(relu_bwd_kernel, dim3(ew_grid(n)), dim3(256), 0, st,
                       (const __bf16*)dy, (const __bf16*)y, (__bf16*)dx, n);
}

void ss_softmax_fwd(const void* x, void* s, int B, int C, hipStream_t st) {
    hipLaunchKernelGGL(softmax_fwd_kernel, dim3(cdiv(B, 4)), dim3(256), 0, st,
                       (const __bf16*)x, (__bf16*)s, B, C);
}

void ss_softmax_bwd(const void* dy, const void* s, void* dx, int B, int C,
                    hipStream_t st) {
    hipLaunchKernelGGL(softmax_bwd_kernel, dim3(cdiv(B, 4)), dim3(256), 0, st,
                       (const __bf16*)dy, (const __bf16*)s, (__bf16*)dx, B, C);
}

void ss_head_mse_bwd(const void* s, const void* t, void* dz, int B, int C,
                     float inv_gb, hipStream_t st) {
    hipLaunchKernelGGL(head_mse_bwd_kernel, dim3(cdiv(B, 4)), dim3(256), 0, st,
                       (const __bf16*)s, (const __bf16*)t, (__bf16*)dz, B, C,
                       inv_gb);
}

void ss_head_xent_bwd(const void* s, const void* t, void* dz, long n,
                      float inv_gb, hipStream_t st) {
    hipLaunchKernelGGL(head_xent_bwd_kernel, dim3(ew_grid(n)), dim3(256), 0,
                       st, (const __bf16*)s, (const __bf16*)t, (__bf16*)dz, n,
                       inv_gb);
}

void ss_row_argmax(const void* x, void* out, int B, int C, hipStream_t st) {
    hipLaunchKernelGGL(row_argmax_kernel, dim3(cdiv(B, 4)), dim3(256), 0, st,
                       (const __bf16*)x, (int*)out, B, C);
}

// ---- blockmap fused optimizers ------------------------------------
// v2 of the fused optimizer kernels: a host-precomputed block→(tensor,
// elem_base) map replaces the per-ELEMENT binary search of the v1
// kernels — the dependent 6-level desc walk serialized every loop
// iteration and held the wide-model sgd_multi at ~0.9 TB/s (8.2 ms/step
// at 523M params; profiles/r02_kernel_stats.md).  Each block owns
// 16384 contiguous elements of ONE tensor; float4/ushort4 vector
// accesses; the only scalar fallback is the grad operand when its
// flat-buffer view is not 16B-aligned (odd bias offsets).
#define SS_OPT_EPB 16384

__global__ __launch_bounds__(256) void sgd_multi2_kernel(
    const long* __restrict__ desc, const long* __restrict__ bmap, float lr,
    float momentum, float weight_decay) {
    const long* d = desc + bmap[(long)blockIdx.x * 2] * 8;
    const long base = bmap[(long)blockIdx.x * 2 + 1];
    const long n = d[4];
    float* __restrict__ master = (float*)d[0];
    const float* __restrict__ grad = (const float*)d[1];
    __bf16* lp = (__bf16*)d[2];
    __bf16* lpt = (__bf16*)d[3];
    float* vel = (float*)d[7];
    const long cols = d[5];
    const long rows = cols ? n / cols : 0;
    const bool ga = ((unsigned long)d[1] & 15) == 0;
#pragma unroll 1
    for (int it = 0; it < SS_OPT_EPB / 1024; ++it) {
        const long e = base + (long)it * 1024 + threadIdx.x * 4;
        if (e >= n) break;
        if (e + 4 <= n) {
            float g4[4];
            if (ga) {
                const float4 gv = *(const float4*)(grad + e);
                g4[0] = gv.x; g4[1] = gv.y; g4[2] = gv.z; g4[3] = gv.w;
            } else {
#pragma unroll
                for (int k = 0; k < 4; ++k) g4[k] = grad[e + k];
            }
            float4 m4 = *(float4*)(master + e);
            float* mv = (float*)&m4;
            float v4[4];
            if (vel) {
                float4 vv = *(float4*)(vel + e);
                v4[0] = vv.x; v4[1] = vv.y; v4[2] = vv.z; v4[3] = vv.w;
            }
            __bf16 out[4];
#pragma unroll
            for (int k = 0; k < 4; ++k) {
                float g = g4[k];
                if (weight_decay != 0.f) g += weight_decay * mv[k];
                if (vel) {
                    g = momentum * v4[k] + g;
                    v4[k] = g;
                }
                mv[k] -= lr * g;
                out[k] = f2bf(mv[k]);
            }
            *(float4*)(master + e) = m4;
            if (vel) {
                float4 vv = {v4[0], v4[1], v4[2], v4[3]};
                *(float4*)(vel + e) = vv;
            }
            if (lp) *(ushort4*)(lp + e) = *(ushort4*)out;
            if (lpt) {
#pragma unroll
                for (int k = 0; k < 4; ++k)
                    lpt[((e + k) % cols) * rows + (e + k) / cols] = out[k];
            }
        } else {
            for (long q = e; q < n; ++q) {
                float g = grad[q];
                if (weight_decay != 0.f) g += weight_decay * master[q];
                if (vel) {
                    g = momentum * vel[q] + g;
                    vel[q] = g;
                }
                const float m = master[q] - lr * g;
                master[q] = m;
                const __bf16 v = f2bf(m);
                if (lp) lp[q] = v;
                if (lpt) lpt[(q % cols) * rows + q / cols] = v;
            }
        }
    }
}

__global__ __launch_bounds__(256) void adamw_multi2_kernel(
    const long* __restrict__ desc, const long* __restrict__ bmap, float lr,
    float beta1, float beta2, float eps, float weight_decay, float inv_bc1,
    float inv_bc2) {
    const long* d = desc + bmap[(long)blockIdx.x * 2] * 9;
    const long base = bmap[(long)blockIdx.x * 2 + 1];
    const long n = d[4];
    float* __restrict__ master = (float*)d[0];
    const float* __restrict__ grad = (const float*)d[1];
    __bf16* lp = (__bf16*)d[2];
    __bf16* lpt = (__bf16*)d[3];
    float* __restrict__ ma = (float*)d[7];
    float* __restrict__ va = (float*)d[8];
    const long cols = d[5];
    const long rows = cols ? n / cols : 0;
    const bool ga = ((unsigned long)d[1] & 15) == 0;
#pragma unroll 1
    for (int it = 0; it < SS_OPT_EPB / 1024; ++it) {
        const long e = base + (long)it * 1024 + threadIdx.x * 4;
        if (e >= n) break;
        if (e + 4 <= n) {
            float g4[4];
            if (ga) {
                const float4 gv = *(const float4*)(grad + e);
                g4[0] = gv.x; g4[1] = gv.y; g4[2] = gv.z; g4[3] = gv.w;
            } else {
#pragma unroll
                for (int k = 0; k < 4; ++k) g4[k] = grad[e + k];
            }
            float4 m4 = *(float4*)(master + e);
            float4 ma4 = *(float4*)(ma + e);
            float4 va4 = *(float4*)(va + e);
            float* mv = (float*)&m4;
            float* mav = (float*)&ma4;
            float* vav = (float*)&va4;
            __bf16 out[4];
#pragma unroll
            for (int k = 0; k < 4; ++k) {
                const float g = g4[k];
                const float m1 = beta1 * mav[k] + (1.f - beta1) * g;
                const float v1 = beta2 * vav[k] + (1.f - beta2) * g * g;
                mav[k] = m1;
                vav[k] = v1;
                float pv = mv[k];
                if (weight_decay != 0.f) pv -= lr * weight_decay * pv;
                pv -= lr * (m1 * inv_bc1) / (sqrtf(v1 * inv_bc2) + eps);
                mv[k] = pv;
                out[k] = f2bf(pv);
            }
            *(float4*)(master + e) = m4;
            *(float4*)(ma + e) = ma4;
            *(float4*)(va + e) = va4;
            if (lp) *(ushort4*)(lp + e) = *(ushort4*)out;
            if (lpt) {
#pragma unroll
                for (int k = 0; k < 4; ++k)
                    lpt[((e + k) % cols) * rows + (e + k) / cols] = out[k];
            }
        } else {
            for (long q = e; q < n; ++q) {
                const float g = grad[q];
                const float m1 = beta1 * ma[q] + (1.f - beta1) * g;
                const float v1 = beta2 * va[q] + (1.f - beta2) * g * g;
                ma[q] = m1;
                va[q] = v1;
                float pv = master[q];
                if (weight_decay != 0.f) pv -= lr * weight_decay * pv;
                pv -= lr * (m1 * inv_bc1) / (sqrtf(v1 * inv_bc2) + eps);
                master[q] = pv;
                const __bf16 v = f2bf(pv);
                if (lp) lp[q] = v;
                if (lpt) lpt[(q % cols) * rows + q / cols] = v;
            }
        }
    }
}

// Tiled bf16 transpose: dst[c][r] = src[r][c], 64x64 LDS tiles with
// padding — replaces the fused optimizers' in-kernel transposed
// emission for LARGE 2-D weights, where the per-element scatter store
// (stride rows*2 B) is uncoalesced and dominated the wide-model step
// (measured: sgd_multi 1.65 ms/step on 8x4096-wide — ~7x the
// coalesced roofline; see profiles/r02_evidence.md).
__global__ __launch_bounds__(256) void transpose_bf16_kernel(
    const __bf16* __restrict__ src, __bf16* __restrict__ dst, int rows,
    int cols) {
    __shared__ __bf16 tile[64][65];
    const int t = threadIdx.x;
    const int r0 = blockIdx.y * 64, c0 = blockIdx.x * 64;
    const int lr = t >> 2;           // 0..63
    const int lc = (t & 3) * 16;     // 0,16,32,48
    // load 64x64 source tile, 16B vectors
    {
        const __bf16* p = src + (long)(r0 + lr) * cols + c0 + lc;
#pragma unroll
        for (int v = 0; v < 2; ++v) {
            bf16x8 x = *(const bf16x8*)(p + v * 8);
#pragma unroll
            for (int e = 0; e < 8; ++e) tile[lr][lc + v * 8 + e] = x[e];
        }
    }
    __syncthreads();
    // write transposed, 16B vectors along dst rows (= source columns)
    {
        __bf16* p = dst + (long)(c0 + lr) * rows + r0 + lc;
#pragma unroll
        for (int v = 0; v < 2; ++v) {
            bf16x8 x;
#pragma unroll
            for (int e = 0; e < 8; ++e) x[e] = tile[lc + v * 8 + e][lr];
            *(bf16x8*)(p + v * 8) = x;
        }
    }
}

__global__ __launch_bounds__(256) void adamw_multi_kernel(
    const long* __restrict__ desc, int ntensors, long total, float lr,
    float beta1, float beta2, float eps, float weight_decay,
    float inv_bc1, float inv_bc2) {
    // desc row (9 longs): [master, grad, lp, lp_t, numel, cols, start,
    // exp_avg, exp_avg_sq].  Decoupled weight decay (AdamW); bias
    // corrections inv_bc1 = 1/(1-beta1^t), inv_bc2 = 1/(1-beta2^t)
    // precomputed on the host.  Re-emits the bf16 compute copy and
    // the transposed bf16 copy, same as sgd_multi_kernel.
    long e = (long)blockIdx.x * 256 + threadIdx.x;
    const long stride = (long)gridDim.x * 256;
    for (; e < total; e += stride) {
        int lo = 0, hi = ntensors - 1;
        while (lo < hi) {
            const int mid = (lo + hi + 1) >> 1;
            if (desc[mid * 9 + 6] <= e) lo = mid;
            else hi = mid - 1;
        }
        const long* d = desc + lo * 9;
        const long i = e - d[6];
        float* master = (float*)d[0];
        const float g = ((const float*)d[1])[i];
        float* ma = (float*)d[7];
        float* va = (float*)d[8];
        const float m1 = beta1 * ma[i] + (1.f - beta1) * g;
        const float v1 = beta2 * va[i] + (1.f - beta2) * g * g;
        ma[i] = m1;
        va[i] = v1;
        float p = master[i];
        if (weight_decay != 0.f) p -= lr * weight_decay * p;
        const float mhat = m1 * inv_bc1;
        const float vhat = v1 * inv_bc2;
        p -= lr * mhat / (sqrtf(vhat) + eps);
        master[i] = p;
        const __bf16 v = f2bf(p);
        if (d[2]) ((__bf16*)d[2])[i] = v;
        if (d[3]) {
            const long cols = d[5];
            const long rows = d[4] / cols;
            ((__bf16*)d[3])[(i % cols) * rows + i / cols] = v;
        }
    }
}

void ss_sgd_multi2(const void* desc, const void* bmap, int nblocks,
                   float lr, float momentum, float weight_decay,
                   hipStream_t st) {
    hipLaunchKernelGGL(sgd_multi2_kernel, dim3(nblocks), dim3(256), 0, st,
                       (const long*)desc, (const long*)bmap, lr, momentum,
                       weight_decay);
}

void ss_adamw_multi2(const void* desc, const void* bmap, int nblocks,
                     float lr, float beta1, float beta2, float eps,
                     float weight_decay, float inv_bc1, float inv_bc2,
                     hipStream_t st) {
    hipLaunchKernelGGL(adamw_multi2_kernel, dim3(nblocks), dim3(256), 0, st,
                       (const long*)desc, (const long*)bmap, lr, beta1,
                       beta2, eps, weight_decay, inv_bc1, inv_bc2);
}

void ss_transpose_bf16(const void* src, void* dst, int rows, int cols,
                       hipStream_t st) {
    dim3 grid(cols / 64, rows / 64);
    hipLaunchKernelGGL(transpose_bf16_kernel, grid, dim3(256), 0, st,
                       (const __bf16*)src, (__bf16*)dst, rows, cols);
}

void ss_adamw_multi(const void* desc, int ntensors, long total, float lr,
                    float beta1, float beta2, float eps, float weight_decay,
                    float inv_bc1, float inv_bc2, hipStream_t st) {
    long blocks = (total + 255) / 256;
    if (blocks > 1024) blocks = 1024;
    hipLaunchKernelGGL(adamw_multi_kernel, dim3((int)blocks), dim3(256), 0,
                       st, (const long*)desc, ntensors, total, lr, beta1,
                       beta2, eps, weight_decay, inv_bc1, inv_bc2);
}

void ss_sgd_multi(const void* desc, int ntensors, long total, float lr,
                  float momentum, float weight_decay, hipStream_t st) {
    long blocks = (total + 255) / 256;
    if (blocks > 1024) blocks = 1024;
    hipLaunchKernelGGL(sgd_multi_kernel, dim3((int)blocks), dim3(256), 0, st,
                       (const long*)desc, ntensors, total, lr, momentum,
                       weight_decay);
}

}  // extern "C"
