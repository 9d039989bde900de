// Elementwise / reduction helpers for the explicit backward path:
// relu_bwd (dx = dy * (act>0)) and colsum (bias gradients).
#include "common.h"

namespace {

__global__ void relu_bwd_kernel(const bf16_t* __restrict__ dy,
                                const bf16_t* __restrict__ act,
                                bf16_t* __restrict__ dx, long n) {
  long i = (long)(blockIdx.x * blockDim.x + threadIdx.x);
  long stride = (long)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    float a = bf2f(act[i]);
    dx[i] = a > 0.f ? dy[i] : f2bf(0.f);
  }
}

// colsum: out[n] = sum_m x[m][n], x bf16 [M,N], out fp32.
// One workgroup per column strip of 256; threads own columns (coalesced
// row-major reads: consecutive lanes read consecutive columns).
__global__ void colsum_kernel(const bf16_t* __restrict__ x,
                              float* __restrict__ out, int M, int N) {
  const int col = blockIdx.x * blockDim.x + threadIdx.x;
  if (col >= N) return;
  float s = 0.f;
  for (int m = 0; m < M; ++m) s += bf2f(x[(long)m * N + col]);
  out[col] = s;
}

// N-way bf16 add: out = sum of up to 6 same-shape tensors in ONE
// pass (autograd's multi-consumer grad fan-in is (n-1) pairwise add
// kernels — 35 of them per Inception step).
struct AddNArgs {
  const bf16_t* src[6];
  int n;
};

__global__ __launch_bounds__(256)
void add_n_kernel(AddNArgs a, bf16_t* __restrict__ out, long total) {
  const long i0 = ((long)blockIdx.x * 256 + threadIdx.x) * 8;
  const long stride = (long)gridDim.x * 256 * 8;
  for (long i = i0; i < total; i += stride) {
    if (i + 8 <= total) {
      f32x4 lo = {}, hi = {};
      for (int t = 0; t < a.n; ++t) {
        const bf16x8 v = *(const bf16x8*)&((const __bf16*)a.src[t])[i];
#pragma unroll
        for (int j = 0; j < 4; ++j) lo[j] += (float)v[j];
#pragma unroll
        for (int j = 0; j < 4; ++j) hi[j] += (float)v[4 + j];
      }
      bf16x8 o;
#pragma unroll
      for (int j = 0; j < 4; ++j) o[j] = (__bf16)lo[j];
#pragma unroll
      for (int j = 0; j < 4; ++j) o[4 + j] = (__bf16)hi[j];
      *(bf16x8*)&((__bf16*)out)[i] = o;
    } else {
      for (long k = i; k < total; ++k) {
        float v = 0.f;
        for (int t = 0; t < a.n; ++t) v += bf2f(a.src[t][k]);
        out[k] = f2bf(v);
      }
    }
  }
}

}  // namespace

void launch_add_n(const bf16_t* const* srcs, int n, bf16_t* out, long total,
                  hipStream_t stream) {
  AddNArgs a = {};
  a.n = n;
  for (int i = 0; i < n; ++i) a.src[i] = srcs[i];
  long blocks = (total / 8 + 255) / 256;
  if (blocks > 4096) blocks = 4096;
  if (blocks < 1) blocks = 1;
  hipLaunchKernelGGL(add_n_kernel, dim3((unsigned)blocks), dim3(256), 0,
                     stream, a, out, total);
}

void launch_relu_bwd(const bf16_t* dy, const bf16_t* act, bf16_t* dx, long n,
                     hipStream_t stream) {
  long blocks = (n + 255) / 256;
  if (blocks > 2048) blocks = 2048;
  hipLaunchKernelGGL(relu_bwd_kernel, dim3((unsigned)blocks), dim3(256), 0,
                     stream, dy, act, dx, n);
}

void launch_colsum(const bf16_t* x, float* out, int M, int N,
                   hipStream_t stream) {
  hipLaunchKernelGGL(colsum_kernel, dim3(ceil_div(N, 256)), dim3(256), 0,
                     stream, x, out, M, N);
}
