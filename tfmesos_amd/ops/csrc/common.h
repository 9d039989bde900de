// Common device helpers for tfmesos_amd CDNA4 (gfx950) kernels.
// Target: MI355X only — wave64, MFMA, 160 KiB LDS/CU. No CUDA compat.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE 64
#define DEVINL __device__ __forceinline__

typedef __hip_bfloat16 bf16_t;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(2))) float f32x2;

DEVINL float bf2f(bf16_t v) { return __bfloat162float(v); }
DEVINL bf16_t f2bf(float v) { return __float2bfloat16(v); }

static inline int ceil_div(int a, int b) { return (a + b - 1) / b; }

// host-side descriptor for gemm_small's fused SGD tail (gemm.hip) —
// opaque pointers so the binding TU needs no device vector types
struct SmallSgdArgs {
  float* pmw; void* psw;      // W master fp32 / bf16 shadow [M,N]
  float* pmb; void* psb;      // bias master / shadow [N]
  const void* g2w;            // classifier W grad (bf16, from the head)
  float* pm2w; void* ps2w;
  const void* g2b;
  float* pm2b; void* ps2b;
  float lr;
  int n2w, n2b;
};

#define HIP_CHECK(expr)                                                     \
  do {                                                                      \
    hipError_t _e = (expr);                                                 \
    if (_e != hipSuccess) {                                                 \
      TORCH_CHECK(false, "HIP error: ", hipGetErrorString(_e), " at ",      \
                  __FILE__, ":", __LINE__);                                 \
    }                                                                       \
  } while (0)
