#include "hip/hip_runtime.h"
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

}  // namespace

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
