// Probe 4: exact k-window of every scale slot (row-0 slots).
// A[row][k] = 1 only for true-k in [h*16, h*16+16) (one 16-window),
// B = all ones.  Patch scale_a byte b of lane L (L&15==0 so row 0)
// to 2.0; D[0][0] - 16 = how many of that window's k got scaled.

#include <hip/hip_runtime.h>
#include <cstdio>

using f32x4 = __attribute__((ext_vector_type(4))) float;
using i32x8 = __attribute__((ext_vector_type(8))) int;

__global__ void probe(float* d_out, int L, int byte, int h, int which) {
    const int lane = threadIdx.x & 63;
    const int row = lane & 15;
    const int kq = lane >> 4;
    i32x8 a{}, b{};
    unsigned char* ab = (unsigned char*)&a;
    unsigned char* bb = (unsigned char*)&b;
    for (int j = 0; j < 32; ++j) {
        const int k = kq * 32 + j;  // true k (probe3 part 1)
        ab[j] = (k >= h * 16 && k < h * 16 + 16) ? 0x38 : 0;
        bb[j] = 0x38;
    }
    int sa = 0x7f7f7f7f, sb = 0x7f7f7f7f;
    int patched =
        (0x7f7f7f7f & ~(0xff << (8 * byte))) | (0x80 << (8 * byte));
    if (lane == L) { if (which == 0) sa = patched; else sb = patched; }
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    acc = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
        a, b, acc, 0, 0, 0, sa, 0, sb);
    for (int r = 0; r < 4; ++r) d_out[(kq * 4 + r) * 16 + row] = acc[r];
}

int main() {
    float* d;
    hipMalloc(&d, 256 * sizeof(float));
    float h_[256];
    for (int which = 0; which < 2; ++which) {
        printf("== %s: rows=k-window h, cols show extra per (L,b) ==\n",
               which ? "scale_b (col 0)" : "scale_a (row 0)");
        printf("         ");
        for (int L = 0; L < 64; L += 16)
            for (int b = 0; b < 4; ++b) printf(" L%02db%d", L, b);
        printf("\n");
        for (int hw = 0; hw < 8; ++hw) {
            printf("k[%3d-%3d)", hw * 16, hw * 16 + 16);
            for (int L = 0; L < 64; L += 16) {
                for (int b = 0; b < 4; ++b) {
                    hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, d, L,
                                       b, hw, which);
                    hipMemcpy(h_, d, sizeof(h_), hipMemcpyDeviceToHost);
                    printf(" %5.0f", h_[0] - 16.f);
                }
            }
            printf("\n");
        }
    }
    return 0;
}
