// Probe: __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4 operand and
// scale layout on gfx950 (the ISA doc is not in this image; the bf16
// MFMA and tr16 layouts were established the same way, probe_tr.hip).
//
//   hipcc --offload-arch=gfx950 scripts/probe_mx.hip -o /tmp/probe_mx
//
// Hypothesis H0 (natural extension of 16x16x32_bf16):
//   A: lane l holds A[l&15][(l>>4)*32 + j], j = byte index 0..31 of
//      the 8 i32s (little-endian, consecutive k)
//   B: lane l holds B[l&15][(l>>4)*32 + j]   (B = N-major rows)
//   C/D: dtype-independent: row=(l>>4)*4+r, col=l&15
//   scale: per-32-block E8M0; lane's block = l>>4; opsel picks the
//      byte of the i32 scale operand.
// The probe fills A/B with exactly-representable e4m3 values by the
// H0 formula, computes D, and compares against a host reference; it
// then re-runs with scale byte 128 (=2.0) on A to verify opsel/byte
// semantics.

#include <hip/hip_runtime.h>
#include <cstdio>
#include <cmath>

using f32x4 = __attribute__((ext_vector_type(4))) float;
using i32x8 = __attribute__((ext_vector_type(8))) int;

// e4m3fn encode for small exact values (powers of two * small ints)
__host__ __device__ unsigned char f2e4m3(float v) {
    // only needs exactness for v in {±0.5,±1,±1.5,±2,±3,±4,±6,±8}
    unsigned char s = v < 0 ? 0x80 : 0;
    float a = fabsf(v);
    if (a == 0.f) return s;
    int e = 0;
    while (a >= 2.f) { a /= 2.f; ++e; }
    while (a < 1.f) { a *= 2.f; --e; }
    int m = (int)roundf((a - 1.f) * 8.f);
    return s | (unsigned char)(((e + 7) << 3) | m);
}

__global__ void probe(float* d_out, int scale_mode) {
    const int lane = threadIdx.x & 63;
    const int row = lane & 15;
    const int kq = lane >> 4;  // k-quarter 0..3
    i32x8 a, b;
    unsigned char* ab = (unsigned char*)&a;
    unsigned char* bb = (unsigned char*)&b;
    for (int j = 0; j < 32; ++j) {
        const int k = kq * 32 + j;
        // A[row][k] = pattern, B[col][k] = pattern (asymmetric!)
        float av = ((row + k) % 7) * 0.5f - 1.5f;
        float bv = ((3 * row + 2 * k) % 5) * 0.5f - 1.f;
        ab[j] = f2e4m3(av);
        bb[j] = f2e4m3(bv);
    }
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    int sa = 0x7f7f7f7f;  // E8M0 127 = x1.0 in every byte
    if (scale_mode == 1) sa = 0x7f7f7f80;  // byte0 = 128 = x2.0
    if (scale_mode == 2) sa = 0x7f807f7f;  // byte2 = 128
    if (scale_mode == 3 && kq == 1) sa = 0x7f7f7f80;   // lane-varying:
    if (scale_mode == 4 && row == 0) sa = 0x7f7f7f80;  // block/row tests
    acc = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
        a, b, acc, 0 /*fmtA=fp8*/, 0 /*fmtB=fp8*/, 0, sa, 0, 0x7f7f7f7f);
    for (int r = 0; r < 4; ++r) {
        const int drow = kq * 4 + r;
        const int dcol = row;
        d_out[drow * 16 + dcol] = acc[r];
    }
}

int main() {
    float* d;
    hipMalloc(&d, 256 * sizeof(float));
    float h[256], ref[256];
    // host reference under H0
    auto aval = [](int r, int k) {
        return (float)(((r + k) % 7) * 0.5f - 1.5f);
    };
    auto bval = [](int c, int k) {
        return (float)(((3 * c + 2 * k) % 5) * 0.5f - 1.f);
    };
    for (int mode = 0; mode < 5; ++mode) {
        hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, d, mode);
        hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
        double maxerr = 0;
        for (int r = 0; r < 16; ++r)
            for (int c = 0; c < 16; ++c) {
                double s = 0;
                for (int k = 0; k < 128; ++k) {
                    double sc = 1.0;
                    // hypothesis for scale application per mode:
                    // mode1: A-scale byte0 doubles... which k-blocks?
                    // leave unscaled here; mismatch pattern shows it
                    s += aval(r, k) * bval(c, k) * sc;
                }
                ref[r * 16 + c] = (float)s;
                maxerr = fmax(maxerr, fabs(h[r * 16 + c] - s));
            }
        printf("mode %d: H0 maxerr=%g\n", mode, maxerr);
        if (mode > 0) {
            // report the D rows/cols whose values changed vs mode-0
            // expectation by exactly 2x on part of the sum
            for (int r = 0; r < 16; ++r) {
                double rowdelta = 0;
                for (int c = 0; c < 16; ++c)
                    rowdelta += h[r * 16 + c] - ref[r * 16 + c];
                if (fabs(rowdelta) > 0.25)
                    printf("  mode %d: D-row %d delta-sum %.2f\n", mode, r,
                           rowdelta);
            }
        }
    }
    // dump one corner for eyeballing
    printf("D[0][0..3] = %g %g %g %g\n", h[0], h[1], h[2], h[3]);
    return 0;
}
