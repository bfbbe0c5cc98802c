// Shared helpers for the shallowspeed_amd HIP/CDNA4 (gfx950) kernels.
//
// Targets MI355X only: wave64, 4 SIMD-32/CU, MFMA bf16 matrix cores,
// 160 KiB LDS/CU.  No CUDA-compat paths, no other archs.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <type_traits>

#define WAVE 64

using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x4 = __attribute__((ext_vector_type(4))) float;

__device__ __forceinline__ float bf2f(__bf16 v) { return (float)v; }
__device__ __forceinline__ __bf16 f2bf(float v) { return (__bf16)v; }

// ceil-div
constexpr int cdiv(int a, int b) { return (a + b - 1) / b; }

#define HIP_CHECK(expr)                                                     \
    do {                                                                    \
        hipError_t _e = (expr);                                             \
        if (_e != hipSuccess) {                                             \
            TORCH_CHECK(false, "HIP error: ", hipGetErrorString(_e));       \
        }                                                                   \
    } while (0)
