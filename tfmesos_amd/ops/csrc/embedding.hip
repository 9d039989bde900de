// Embedding gather / scatter-add — the sparse PS push/pull path.
//
// The reference's PS architecture implies sparse row push/pull (row-factor
// model in examples/matrix_factorization.py:21-28; BASELINE.json names a
// "sparse-embedding PS ... sparse push/pull HIP path"). One wave64 per
// row; gather streams rows out (bf16 or fp32 table), scatter-add
// accumulates fp32 gradients with atomics (duplicate ids correct).
#include "common.h"

namespace {

template <typename T>
__global__ void gather_kernel(const T* __restrict__ table,
                              const long* __restrict__ ids,
                              T* __restrict__ out, int D, long V) {
  const long row = blockIdx.x;
  const long id = ids[row];
  const T* src = table + id * D;
  T* dst = out + row * D;
  if (id < 0 || id >= V) {  // defensive: zero-fill out-of-range ids
    for (int c = threadIdx.x; c < D; c += WAVE) dst[c] = (T)0.f;
    return;
  }
  for (int c = threadIdx.x; c < D; c += WAVE) dst[c] = src[c];
}

template <typename G>
__global__ void scatter_add_kernel(float* __restrict__ table,
                                   const long* __restrict__ ids,
                                   const G* __restrict__ rows, int D, long V) {
  const long row = blockIdx.x;
  const long id = ids[row];
  if (id < 0 || id >= V) return;
  const G* src = rows + row * D;
  float* dst = table + id * D;
  for (int c = threadIdx.x; c < D; c += WAVE) {
    float v;
    if constexpr (sizeof(G) == 2) v = bf2f(*(const bf16_t*)&src[c]);
    else v = *(const float*)&src[c];
    atomicAdd(&dst[c], v);
  }
}

}  // namespace

void launch_gather_bf16(const bf16_t* table, const long* ids, bf16_t* out,
                        long n, int D, long V, hipStream_t stream) {
  hipLaunchKernelGGL(gather_kernel<bf16_t>, dim3((unsigned)n), dim3(WAVE), 0,
                     stream, table, ids, out, D, V);
}

void launch_gather_f32(const float* table, const long* ids, float* out,
                       long n, int D, long V, hipStream_t stream) {
  hipLaunchKernelGGL(gather_kernel<float>, dim3((unsigned)n), dim3(WAVE), 0,
                     stream, table, ids, out, D, V);
}

void launch_scatter_add_bf16(float* table, const long* ids,
                             const bf16_t* rows, long n, int D, long V,
                             hipStream_t stream) {
  hipLaunchKernelGGL(scatter_add_kernel<bf16_t>, dim3((unsigned)n),
                     dim3(WAVE), 0, stream, table, ids, rows, D, V);
}

void launch_scatter_add_f32(float* table, const long* ids, const float* rows,
                            long n, int D, long V, hipStream_t stream) {
  hipLaunchKernelGGL(scatter_add_kernel<float>, dim3((unsigned)n),
                     dim3(WAVE), 0, stream, table, ids, rows, D, V);
}
