// Probe 3 (decisive): exact k-space layout + scale-slot partition of
// mfma_scale_f32_16x16x128_f8f6f4.
//
// Part 1 — data k-order: A[lane byte j] = j (as e4m3-exact values
// 0..31 scaled into exact range), B = delta at one true-k k0 probed
// by putting delta in ONE byte of ONE lane-group; D[r][c] reads back
// which (lane-group, byte) of A contributes at the same true k.
// Simpler equivalent: set B[lane l', byte j'] = 1 for exactly one
// (group g', byte j') per launch and A = byte-index everywhere; the
// contraction picks out A bytes sharing true-k with that B byte:
// D[r][c] = A-value at that k = the A byte index holding it.
//
// Part 2 — scale partition: A = B = all ones; for each (lane L, byte
// b): patch scale_a to 2.0 there; D[r][*] - 128 = # of row-r true-k
// positions scaled.  Reported per (L, b): affected row + count.

#include <hip/hip_runtime.h>
#include <cstdio>

using f32x4 = __attribute__((ext_vector_type(4))) float;
using i32x8 = __attribute__((ext_vector_type(8))) int;

__device__ __host__ unsigned char enc(int v) {
    // exact e4m3 for 0..31: v<16 -> v/2 grid... use v in {0..15} via
    // 1..15 exactly representable? e4m3 ints exact up to 16, then
    // evens to 32.  encode value v exactly for v in 0..15.
    if (v == 0) return 0;
    float a = (float)v;
    int e = 0;
    while (a >= 2.f) { a /= 2.f; ++e; }
    int m = (int)((a - 1.f) * 8.f + 0.5f);
    return (unsigned char)(((e + 7) << 3) | m);
}

__device__ unsigned char encj(int j) {
    // 32 distinct exact e4m3 values identifying byte index j:
    // j 0-7: 1+j/8; 8-15: 2+(j-8)/4; 16-23: 4+(j-16)/2; 24-31: 8+(j-24)
    int seg = j >> 3, off = j & 7;
    int e = seg;             // exponent 0..3 -> 1,2,4,8
    int m = off;             // mantissa eighths
    return (unsigned char)(((e + 7) << 3) | m);
}

__global__ void k_order(float* d_out, int gsel, int jsel, int pass) {
    const int lane = threadIdx.x & 63;
    const int row = lane & 15;
    const int kq = lane >> 4;
    i32x8 a{}, b{};
    unsigned char* ab = (unsigned char*)&a;
    unsigned char* bb = (unsigned char*)&b;
    for (int j = 0; j < 32; ++j) {
        // pass 0: value identifies byte index j; pass 1: lane group
        ab[j] = pass == 0 ? encj(j) : enc(1 + kq);
        bb[j] = 0;
    }
    if (kq == gsel) bb[jsel] = 0x38;  // 1.0 at (group gsel, byte jsel)
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    acc = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
        a, b, acc, 0, 0, 0, 0x7f7f7f7f, 0, 0x7f7f7f7f);
    for (int r = 0; r < 4; ++r) d_out[(kq * 4 + r) * 16 + row] = acc[r];
}

__global__ void scale_map(float* d_out, int L, int byte) {
    const int lane = threadIdx.x & 63;
    const int row = lane & 15;
    const int kq = lane >> 4;
    i32x8 a, b;
    unsigned char* ab = (unsigned char*)&a;
    unsigned char* bb = (unsigned char*)&b;
    for (int j = 0; j < 32; ++j) { ab[j] = 0x38; bb[j] = 0x38; }
    int sa = 0x7f7f7f7f;
    if (lane == L)
        sa = (0x7f7f7f7f & ~(0xff << (8 * byte))) | (0x80 << (8 * byte));
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    acc = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
        a, b, acc, 0, 0, 0, sa, 0, 0x7f7f7f7f);
    for (int r = 0; r < 4; ++r) d_out[(kq * 4 + r) * 16 + row] = acc[r];
}

int main() {
    float* d;
    hipMalloc(&d, 256 * sizeof(float));
    float h[256];
    // Part 1: scan a few (gsel, jsel): value at D[0][0] = 1+(jA&7)
    // where jA is A's byte sharing true-k with B byte (gsel,jsel).
    printf("== k-order: D[0][0] for B-delta at (g,j) ==\n");
    for (int g = 0; g < 4; ++g) {
        for (int pass = 0; pass < 2; ++pass) {
            printf("g=%d p%d:", g, pass);
            for (int j = 0; j < 32; ++j) {
                hipLaunchKernelGGL(k_order, dim3(1), dim3(64), 0, 0, d, g, j,
                                   pass);
                hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
                printf(" %5.3f", h[0]);
            }
            printf("\n");
        }
    }
    // Part 2: full (L, byte) scale map
    printf("== scale map: (L,b) -> row, extra (= #k scaled) ==\n");
    for (int L = 0; L < 64; ++L) {
        for (int b = 0; b < 4; ++b) {
            hipLaunchKernelGGL(scale_map, dim3(1), dim3(64), 0, 0, d, L, b);
            hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
            for (int r = 0; r < 16; ++r)
                if (h[r * 16 + 0] > 128.5f)
                    printf("L=%2d b=%d -> row %2d +%3.0f\n", L, b, r,
                           h[r * 16 + 0] - 128.f);
        }
    }
    return 0;
}
