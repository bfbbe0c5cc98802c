// Empirical probe of gfx950 ds_read_b64_tr_b16 semantics.
// Fills LDS with ushort value = element index, reads via the builtin
// with a candidate per-lane address pattern, prints lane->elements map.
#include <hip/hip_runtime.h>
#include <cstdio>

using u16x4 = __attribute__((ext_vector_type(4))) short;
typedef __attribute__((address_space(3))) u16x4 lds_u16x4;

__global__ void probe(int pattern, ushort* out /* [64][4] */) {
    __shared__ ushort lds[256];
    const int l = threadIdx.x;
    for (int i = l; i < 256; i += 64) lds[i] = (ushort)i;
    __syncthreads();
    int addr_elems = 0;
    switch (pattern) {
        case 0: addr_elems = (l & 15) * 4; break;          // 8B per lane, contiguous by lane
        case 1: addr_elems = (l & 15) * 4 + (l >> 4) * 64; break;
        case 2: addr_elems = (l & 15);     break;          // 2B steps (likely misaligned)
        case 3: addr_elems = (l & 3) * 16 + ((l >> 2) & 3) * 4; break; // row-of-4x16 halves
        case 4: addr_elems = (l & 3) * 16 + ((l >> 2) & 3) * 4 + (l >> 4) * 64; break;
    }
    u16x4 v = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
        (lds_u16x4*)&lds[addr_elems]);
    for (int j = 0; j < 4; ++j) out[l * 4 + j] = (ushort)v[j];
}

int main() {
    ushort* d;
    hipMalloc(&d, 64 * 4 * sizeof(ushort));
    ushort h[256];
    for (int p = 0; p <= 4; ++p) {
        hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, p, d);
        hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
        printf("pattern %d:\n", p);
        for (int l = 0; l < 64; ++l) {
            printf("  lane %2d: %3d %3d %3d %3d\n", l, h[l * 4], h[l * 4 + 1],
                   h[l * 4 + 2], h[l * 4 + 3]);
            if ((l & 15) == 15) l += 0;  // print all
        }
    }
    hipFree(d);
    return 0;
}
