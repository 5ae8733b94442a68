// Common types / helpers for the waternet_amd CDNA4 (gfx950) kernel library.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <cstdint>

// MFMA fragment vector types (gfx950: mfma_f32_16x16x32_bf16 takes <8 x bf16>,
// accumulates <4 x float>)
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(16))) float f32x16;
typedef __attribute__((ext_vector_type(2))) float f32x2;
typedef __attribute__((ext_vector_type(4))) std::uint32_t u32x4;

using bf16_t = __bf16;

#define WN_DEVFN __device__ __forceinline__

constexpr int WAVE = 64;

WN_DEVFN float bf2f(bf16_t v) { return static_cast<float>(v); }
WN_DEVFN bf16_t f2bf(float v) { return static_cast<bf16_t>(v); }

static inline int ceil_div(int a, int b) { return (a + b - 1) / b; }

// Activation codes shared between HIP kernels and the Python bindings
enum ActKind : int {
  ACT_NONE = 0,
  ACT_RELU = 1,
  ACT_SIGMOID = 2,
};

#define HIP_CHECK_LAST()                                                     \
  do {                                                                       \
    hipError_t _e = hipGetLastError();                                       \
    if (_e != hipSuccess) {                                                  \
      TORCH_CHECK(false, "HIP kernel launch failed: ",                       \
                  hipGetErrorString(_e));                                    \
    }                                                                        \
  } while (0)
