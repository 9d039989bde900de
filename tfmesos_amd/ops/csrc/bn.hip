// Fused train-mode batch-norm (+ReLU) for NCHW bf16 activations.
//
// The Inception blocks are conv -> BN -> relu; profiling the torch
// fallback (profiles/r01_inception_n1_kernel_stats.txt) showed the BN
// tensor-math soup (fp32 casts + separate mean/var/normalize/affine/relu
// kernels) at ~36% of the training step. Here each direction is three
// hand-written kernels, all bf16-in/bf16-out with fp32 math:
//
//   fwd: stats-partial (grid C x Z, plane-coalesced reduction, no
//        atomics -> per-slice partials) -> finalize (mean/invstd) ->
//        apply (normalize+affine+ReLU fused, one pass)
//   bwd: stats-partial (sum dy_eff, sum dy_eff*xhat; the ReLU mask
//        y>0 is fused in) -> finalize (dgamma/dbeta + normalized sums)
//        -> apply (dx in one pass)
//
// Reductions read each element once; scalar loads coalesce across the
// 64 lanes (planes are contiguous), so the kernels are bandwidth-bound.
#include "common.h"

namespace {

__global__ __launch_bounds__(256)
void bn_stats_kernel(const __bf16* __restrict__ x, float* __restrict__ part,
                     int N, int C, long HW, int Z) {
  const int c = blockIdx.x;
  const int z = blockIdx.y;
  const int t = threadIdx.x;
  float sum = 0.f, sq = 0.f;
  for (int n = 0; n < N; ++n) {
    const __bf16* plane = x + ((long)n * C + c) * HW;
    for (long i = (long)z * 256 + t; i < HW; i += (long)Z * 256) {
      const float v = (float)plane[i];
      sum += v;
      sq += v * v;
    }
  }
  __shared__ float ls[256], lq[256];
  ls[t] = sum;
  lq[t] = sq;
  __syncthreads();
#pragma unroll
  for (int s = 128; s > 0; s >>= 1) {
    if (t < s) { ls[t] += ls[t + s]; lq[t] += lq[t + s]; }
    __syncthreads();
  }
  if (t == 0) {
    part[((long)z * C + c) * 2] = ls[0];
    part[((long)z * C + c) * 2 + 1] = lq[0];
  }
}

__global__ void bn_finalize_kernel(const float* __restrict__ part,
                                   float* __restrict__ mean,
                                   float* __restrict__ invstd,
                                   int C, int Z, float inv_count, float eps) {
  const int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float s = 0.f, q = 0.f;
  for (int z = 0; z < Z; ++z) {
    s += part[((long)z * C + c) * 2];
    q += part[((long)z * C + c) * 2 + 1];
  }
  const float m = s * inv_count;
  float var = q * inv_count - m * m;
  if (var < 0.f) var = 0.f;
  mean[c] = m;
  invstd[c] = __frsqrt_rn(var + eps);
}

template <bool RELU>
__global__ __launch_bounds__(256)
void bn_apply_kernel(const __bf16* __restrict__ x,
                     const float* __restrict__ mean,
                     const float* __restrict__ invstd,
                     const float* __restrict__ g, const float* __restrict__ b,
                     __bf16* __restrict__ y, int C, long HW) {
  const long p = blockIdx.x;           // plane n*C + c
  const int c = (int)(p % C);
  const float mu = mean[c], is = invstd[c];
  const float sc = g[c] * is, sh = b[c] - mu * sc;
  const __bf16* xin = x + p * HW;
  __bf16* yout = y + p * HW;
  for (long i = threadIdx.x; i < HW; i += 256) {
    float v = (float)xin[i] * sc + sh;
    if (RELU) v = v > 0.f ? v : 0.f;
    yout[i] = (__bf16)v;
  }
}

template <bool RELU>
__global__ __launch_bounds__(256)
void bn_bwd_stats_kernel(const __bf16* __restrict__ x,
                         const __bf16* __restrict__ dy,
                         const __bf16* __restrict__ y,
                         const float* __restrict__ mean,
                         const float* __restrict__ invstd,
                         float* __restrict__ part, int N, int C, long HW,
                         int Z) {
  const int c = blockIdx.x;
  const int z = blockIdx.y;
  const int t = threadIdx.x;
  const float mu = mean[c], is = invstd[c];
  float s1 = 0.f, s2 = 0.f;
  for (int n = 0; n < N; ++n) {
    const long base = ((long)n * C + c) * HW;
    for (long i = (long)z * 256 + t; i < HW; i += (long)Z * 256) {
      float d = (float)dy[base + i];
      if (RELU && (float)y[base + i] <= 0.f) d = 0.f;
      const float xh = ((float)x[base + i] - mu) * is;
      s1 += d;
      s2 += d * xh;
    }
  }
  __shared__ float l1[256], l2[256];
  l1[t] = s1;
  l2[t] = s2;
  __syncthreads();
#pragma unroll
  for (int s = 128; s > 0; s >>= 1) {
    if (t < s) { l1[t] += l1[t + s]; l2[t] += l2[t + s]; }
    __syncthreads();
  }
  if (t == 0) {
    part[((long)z * C + c) * 2] = l1[0];
    part[((long)z * C + c) * 2 + 1] = l2[0];
  }
}

__global__ void bn_bwd_finalize_kernel(const float* __restrict__ part,
                                       float* __restrict__ dgamma,
                                       float* __restrict__ dbeta,
                                       float* __restrict__ s1n,
                                       float* __restrict__ s2n,
                                       int C, int Z, float inv_count) {
  const int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float s1 = 0.f, s2 = 0.f;
  for (int z = 0; z < Z; ++z) {
    s1 += part[((long)z * C + c) * 2];
    s2 += part[((long)z * C + c) * 2 + 1];
  }
  dbeta[c] = s1;
  dgamma[c] = s2;
  s1n[c] = s1 * inv_count;
  s2n[c] = s2 * inv_count;
}

template <bool RELU>
__global__ __launch_bounds__(256)
void bn_bwd_apply_kernel(const __bf16* __restrict__ x,
                         const __bf16* __restrict__ dy,
                         const __bf16* __restrict__ y,
                         const float* __restrict__ mean,
                         const float* __restrict__ invstd,
                         const float* __restrict__ g,
                         const float* __restrict__ s1n,
                         const float* __restrict__ s2n,
                         __bf16* __restrict__ dx, int C, long HW) {
  const long p = blockIdx.x;
  const int c = (int)(p % C);
  const float mu = mean[c], is = invstd[c];
  const float gs = g[c] * is;
  const float a = s1n[c], bb = s2n[c];
  const long base = p * HW;
  for (long i = threadIdx.x; i < HW; i += 256) {
    float d = (float)dy[base + i];
    if (RELU && (float)y[base + i] <= 0.f) d = 0.f;
    const float xh = ((float)x[base + i] - mu) * is;
    dx[base + i] = (__bf16)(gs * (d - a - xh * bb));
  }
}

inline int stats_slices(int N, int C, long HW) {
  // target >=1024 workgroups across the C x Z grid, but keep each
  // slice >=4 round-trips of 256 threads
  long per = ((long)N * HW) / (256 * 4);
  long want = (1024 + C - 1) / C;
  long z = want < per ? want : per;
  if (z < 1) z = 1;
  if (z > 64) z = 64;
  return (int)z;
}

}  // namespace

void launch_bn_fwd(const bf16_t* x, const float* g, const float* b,
                   bf16_t* y, float* mean, float* invstd, float* part,
                   int N, int C, long HW, int Z, float eps, bool relu,
                   hipStream_t stream) {
  dim3 sg(C, Z), sb(256);
  hipLaunchKernelGGL(bn_stats_kernel, sg, sb, 0, stream, (const __bf16*)x,
                     part, N, C, HW, Z);
  hipLaunchKernelGGL(bn_finalize_kernel, dim3(ceil_div(C, 256)), dim3(256),
                     0, stream, part, mean, invstd, C, Z,
                     1.f / ((float)N * HW), eps);
  dim3 ag((unsigned)((long)N * C)), ab(256);
  if (relu)
    hipLaunchKernelGGL((bn_apply_kernel<true>), ag, ab, 0, stream,
                       (const __bf16*)x, mean, invstd, g, b, (__bf16*)y, C, HW);
  else
    hipLaunchKernelGGL((bn_apply_kernel<false>), ag, ab, 0, stream,
                       (const __bf16*)x, mean, invstd, g, b, (__bf16*)y, C, HW);
}

void launch_bn_bwd(const bf16_t* x, const bf16_t* dy, const bf16_t* y,
                   const float* g, const float* mean, const float* invstd,
                   bf16_t* dx, float* dgamma, float* dbeta, float* part,
                   float* s1n, float* s2n, int N, int C, long HW, int Z,
                   bool relu, hipStream_t stream) {
  dim3 sg(C, Z), sb(256);
  if (relu)
    hipLaunchKernelGGL((bn_bwd_stats_kernel<true>), sg, sb, 0, stream,
                       (const __bf16*)x, (const __bf16*)dy, (const __bf16*)y,
                       mean, invstd, part, N, C, HW, Z);
  else
    hipLaunchKernelGGL((bn_bwd_stats_kernel<false>), sg, sb, 0, stream,
                       (const __bf16*)x, (const __bf16*)dy, (const __bf16*)y,
                       mean, invstd, part, N, C, HW, Z);
  hipLaunchKernelGGL(bn_bwd_finalize_kernel, dim3(ceil_div(C, 256)),
                     dim3(256), 0, stream, part, dgamma, dbeta, s1n, s2n, C,
                     Z, 1.f / ((float)N * HW));
  dim3 ag((unsigned)((long)N * C)), ab(256);
  if (relu)
    hipLaunchKernelGGL((bn_bwd_apply_kernel<true>), ag, ab, 0, stream,
                       (const __bf16*)x, (const __bf16*)dy, (const __bf16*)y,
                       mean, invstd, g, s1n, s2n, (__bf16*)dx, C, HW);
  else
    hipLaunchKernelGGL((bn_bwd_apply_kernel<false>), ag, ab, 0, stream,
                       (const __bf16*)x, (const __bf16*)dy, (const __bf16*)y,
                       mean, invstd, g, s1n, s2n, (__bf16*)dx, C, HW);
}

int bn_stats_slices(int N, int C, long HW) { return stats_slices(N, C, HW); }
