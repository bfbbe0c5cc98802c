// Probe 2b: scale-byte sourcing of mfma_scale_f32_16x16x128_f8f6f4.
// A = ones in k-block t only, B = all ones; patch scale_a byte `byte`
// to 2.0 on a lane SUBSET chosen by mode; report affected D cells.

#include <hip/hip_runtime.h>
#include <cstdio>

using f32x4 = __attribute__((ext_vector_type(4))) float;
using i32x8 = __attribute__((ext_vector_type(8))) int;

#define CHK(x) do { hipError_t e_=(x); if (e_) { printf("HIPERR %s\n", hipGetErrorString(e_)); return; } } while (0)

__global__ void probe(float* d_out, int mode, int byte, int t, int which) {
    const int lane = threadIdx.x & 63;
    const int row = lane & 15;
    const int kq = lane >> 4;
    i32x8 a, b;
    unsigned char* ab = (unsigned char*)&a;
    unsigned char* bb = (unsigned char*)&b;
    for (int j = 0; j < 32; ++j) {
        ab[j] = (kq == t) ? 0x38 : 0;
        bb[j] = 0x38;
    }
    bool in = false;
    if (mode == 99) in = true;                       // all lanes
    else if (mode >= 0 && mode < 6) in = (lane >> mode) & 1;  // bit set
    else if (mode >= 100) in = lane == (mode - 100);          // single
    int sa = 0x7f7f7f7f, sb = 0x7f7f7f7f;
    int patched = (0x7f7f7f7f & ~(0xff << (8 * byte))) | (0x80 << (8 * byte));
    if (in) { if (which == 0) sa = patched; else sb = patched; }
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    acc = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
        a, b, acc, 0, 0, 0, sa, 0, sb);
    for (int r = 0; r < 4; ++r) d_out[(kq * 4 + r) * 16 + row] = acc[r];
}

void run(float* d, int mode, int byte, int t, int which) {
    hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, d, mode, byte, t,
                       which);
    CHK(hipGetLastError());
    float h[256];
    CHK(hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost));
    int rows = 0, colsm = 0, nhit = 0;
    for (int r = 0; r < 16; ++r)
        for (int c = 0; c < 16; ++c)
            if (h[r * 16 + c] > 48.f) { rows |= 1 << r; colsm |= 1 << c; ++nhit; }
    printf("%s mode=%3d byte=%d blk=%d base=%g hits=%3d rows=%04x cols=%04x\n",
           which ? "SB" : "SA", mode, byte, t, h[0], nhit, rows, colsm);
}

int main() {
    float* d;
    hipMalloc(&d, 256 * sizeof(float));
    // sanity: all lanes, byte 0, each block
    for (int t = 0; t < 4; ++t) run(d, 99, 0, t, 0);
    // lane-bit subsets, byte 0, block 0 and 1
    for (int bit = 0; bit < 6; ++bit)
        for (int t = 0; t < 2; ++t) run(d, bit, 0, t, 0);
    // bytes 1-3, all lanes
    for (int byte = 1; byte < 4; ++byte) run(d, 99, byte, 0, 0);
    // scale_b all lanes
    for (int t = 0; t < 2; ++t) run(d, 99, 0, t, 1);
    // singles again with sanity-backed machinery
    for (int L : {0, 16, 32, 48}) run(d, 100 + L, 0, 0, 0);
    return 0;
}
