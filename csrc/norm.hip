// LayerNorm + GELU kernels (gfx950, MI355X) — modern-MLP extensions
// beyond the reference's ReLU/Softmax module set.
//
// LayerNorm: wave-per-row, f32 statistics, bf16 I/O.
//   fwd: y = (x - µ) * rstd * γ + β ; saves µ, rstd per row.
//   bwd (input grad): dx = rstd * (g - mean(g) - x̂ * mean(g * x̂)),
//       g = dy * γ, x̂ = (x - µ) * rstd  — wave-per-row.
//   bwd (param grads): dγ[c] += Σ_rows dy * x̂ ; dβ[c] += Σ_rows dy —
//       column reduction, coalesced row-major walk + f32 atomics.
// GELU (tanh approximation): elementwise, bf16 I/O vectorized.
//   bwd needs the PRE-activation z (stashed by the layer).

#include "common.h"

__device__ __forceinline__ float wsum(float v) {
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
    return v;
}

// --------------------------------------------------------------- layernorm

__global__ __launch_bounds__(256) void ln_fwd_kernel(
    const __bf16* __restrict__ x, const __bf16* __restrict__ gamma,
    const __bf16* __restrict__ beta, __bf16* __restrict__ y,
    float* __restrict__ mean, float* __restrict__ rstd, int B, int C,
    float eps) {
    const int lane = threadIdx.x & 63;
    const int row = blockIdx.x * 4 + (threadIdx.x >> 6);
    if (row >= B) return;
    const __bf16* xr = x + (long)row * C;
    float s = 0.f, s2 = 0.f;
    for (int c = lane; c < C; c += 64) {
        const float v = bf2f(xr[c]);
        s += v;
        s2 += v * v;
    }
    s = wsum(s);
    s2 = wsum(s2);
    const float mu = s / C;
    const float var = s2 / C - mu * mu;
    const float rs = rsqrtf(var + eps);
    if (lane == 0) {
        mean[row] = mu;
        rstd[row] = rs;
    }
    __bf16* yr = y + (long)row * C;
    for (int c = lane; c < C; c += 64) {
        const float xh = (bf2f(xr[c]) - mu) * rs;
        yr[c] = f2bf(xh * bf2f(gamma[c]) + bf2f(beta[c]));
    }
}

__global__ __launch_bounds__(256) void ln_bwd_dx_kernel(
    const __bf16* __restrict__ dy, const __bf16* __restrict__ x,
    const __bf16* __restrict__ gamma, const float* __restrict__ mean,
    const float* __restrict__ rstd, __bf16* __restrict__ dx, int B, int C) {
    const int lane = threadIdx.x & 63;
    const int row = blockIdx.x * 4 + (threadIdx.x >> 6);
    if (row >= B) return;
    const __bf16* dyr = dy + (long)row * C;
    const __bf16* xr = x + (long)row * C;
    const float mu = mean[row], rs = rstd[row];
    float sg = 0.f, sgx = 0.f;
    for (int c = lane; c < C; c += 64) {
        const float g = bf2f(dyr[c]) * bf2f(gamma[c]);
        const float xh = (bf2f(xr[c]) - mu) * rs;
        sg += g;
        sgx += g * xh;
    }
    sg = wsum(sg) / C;
    sgx = wsum(sgx) / C;
    __bf16* dxr = dx + (long)row * C;
    for (int c = lane; c < C; c += 64) {
        const float g = bf2f(dyr[c]) * bf2f(gamma[c]);
        const float xh = (bf2f(xr[c]) - mu) * rs;
        dxr[c] = f2bf(rs * (g - sg - xh * sgx));
    }
}

__global__ __launch_bounds__(256) void ln_bwd_dparam_kernel(
    const __bf16* __restrict__ dy, const __bf16* __restrict__ x,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    float* __restrict__ dgamma, float* __restrict__ dbeta, int B, int C,
    int rows_per_block) {
    const int col = blockIdx.x * 256 + threadIdx.x;
    const int r0 = blockIdx.y * rows_per_block;
    const int r1 = min(B, r0 + rows_per_block);
    if (col >= C) return;
    float sg = 0.f, sb = 0.f;
    for (int r = r0; r < r1; ++r) {
        const float d = bf2f(dy[(long)r * C + col]);
        const float xh = (bf2f(x[(long)r * C + col]) - mean[r]) * rstd[r];
        sg += d * xh;
        sb += d;
    }
    atomicAdd(&dgamma[col], sg);
    atomicAdd(&dbeta[col], sb);
}

// ------------------------------------------------------------------ gelu

__device__ __forceinline__ float gelu_f(float z) {
    // tanh approximation
    const float c = 0.7978845608028654f;  // sqrt(2/pi)
    const float u = c * (z + 0.044715f * z * z * z);
    return 0.5f * z * (1.f + tanhf(u));
}

__device__ __forceinline__ float gelu_grad_f(float z) {
    const float c = 0.7978845608028654f;
    const float u = c * (z + 0.044715f * z * z * z);
    const float t = tanhf(u);
    const float du = c * (1.f + 3.f * 0.044715f * z * z);
    return 0.5f * (1.f + t) + 0.5f * z * (1.f - t * t) * du;
}

__global__ __launch_bounds__(256) void gelu_fwd_kernel(
    const __bf16* __restrict__ z, __bf16* __restrict__ y, long n) {
    long i = (long)(blockIdx.x * 256 + threadIdx.x) * 8;
    const long stride = (long)gridDim.x * 256 * 8;
    for (; i + 8 <= n; i += stride) {
        bf16x8 v = *(const bf16x8*)(z + i);
        bf16x8 o;
#pragma unroll
        for (int j = 0; j < 8; ++j) o[j] = f2bf(gelu_f(bf2f(v[j])));
        *(bf16x8*)(y + i) = o;
    }
    if (i < n)
        for (; i < n; ++i) y[i] = f2bf(gelu_f(bf2f(z[i])));
}

__global__ __launch_bounds__(256) void gelu_bwd_kernel(
    const __bf16* __restrict__ dy, const __bf16* __restrict__ z,
    __bf16* __restrict__ dz, long n) {
    long i = (long)(blockIdx.x * 256 + threadIdx.x) * 8;
    const long stride = (long)gridDim.x * 256 * 8;
    for (; i + 8 <= n; i += stride) {
        bf16x8 g = *(const bf16x8*)(dy + i);
        bf16x8 v = *(const bf16x8*)(z + i);
        bf16x8 o;
#pragma unroll
        for (int j = 0; j < 8; ++j)
            o[j] = f2bf(bf2f(g[j]) * gelu_grad_f(bf2f(v[j])));
        *(bf16x8*)(dz + i) = o;
    }
    if (i < n)
        for (; i < n; ++i) dz[i] = f2bf(bf2f(dy[i]) * gelu_grad_f(bf2f(z[i])));
}

// ---------------------------------------------------------------- launchers

extern "C" {

static inline int ngrid(long n) {
    long b = (n / 8 + 255) / 256;
    if (b < 1) b = 1;
    if (b > 2048) b = 2048;
    return (int)b;
}

void ss_ln_fwd(const void* x, const void* gamma, const void* beta, void* y,
               void* mean, void* rstd, int B, int C, float eps,
               hipStream_t st) {
    hipLaunchKernelGGL(ln_fwd_kernel, dim3(cdiv(B, 4)), dim3(256), 0, st,
                       (const __bf16*)x, (const __bf16*)gamma,
                       (const __bf16*)beta, (__bf16*)y, (float*)mean,
                       (float*)rstd, B, C, eps);
}

void ss_ln_bwd_dx(const void* dy, const void* x, const void* gamma,
                  const void* mean, const void* rstd, void* dx, int B, int C,
                  hipStream_t st) {
    hipLaunchKernelGGL(ln_bwd_dx_kernel, dim3(cdiv(B, 4)), dim3(256), 0, st,
                       (const __bf16*)dy, (const __bf16*)x,
                       (const __bf16*)gamma, (const float*)mean,
                       (const float*)rstd, (__bf16*)dx, B, C);
}

void ss_ln_bwd_dparam(const void* dy, const void* x, const void* mean,
                      const void* rstd, void* dgamma, void* dbeta, int B,
                      int C, hipStream_t st) {
    int col_tiles = cdiv(C, 256);
    int splits = cdiv(512, col_tiles);
    if (splits > cdiv(B, 64)) splits = cdiv(B, 64);
    if (splits < 1) splits = 1;
    dim3 grid(col_tiles, splits);
    hipLaunchKernelGGL(ln_bwd_dparam_kernel, grid, dim3(256), 0, st,
                       (const __bf16*)dy, (const __bf16*)x,
                       (const float*)mean, (const float*)rstd, (float*)dgamma,
                       (float*)dbeta, B, C, cdiv(B, splits));
}

void ss_gelu_fwd(const void* z, void* y, long n, hipStream_t st) {
    hipLaunchKernelGGL(gelu_fwd_kernel, dim3(ngrid(n)), dim3(256), 0, st,
                       (const __bf16*)z, (__bf16*)y, n);
}

void ss_gelu_bwd(const void* dy, const void* z, void* dz, long n,
                 hipStream_t st) {
    hipLaunchKernelGGL(gelu_bwd_kernel, dim3(ngrid(n)), dim3(256), 0, st,
                       (const __bf16*)dy, (const __bf16*)z, (__bf16*)dz, n);
}

}  // extern "C"
