// Fused train-mode batch-norm (+ReLU) for channels-last bf16 activations.
//
// The Inception blocks are conv -> BN -> relu; profiling the torch
// fallback (profiles/r01_inception_n1_kernel_stats.txt) showed the BN
// tensor-math soup (fp32 casts + separate mean/var/normalize/affine/relu
// kernels) at ~36% of the training step. The activation layout is the
// conv kernels' channels-last ([P, C] with P = N*H*W pixels, channels
// innermost), so every wave reads 64 consecutive channels — perfectly
// coalesced — and per-channel scale/shift live in LDS for the
// elementwise passes. Three kernels per direction:
//
//   fwd: stats-partial (grid C/64 x Z, no atomics -> per-slice partials)
//        -> finalize (mean/invstd) -> apply (normalize+affine+ReLU, one
//        pass, b128 vectorized when C % 8 == 0)
//   bwd: stats-partial (sum dy_eff, sum dy_eff*xhat; ReLU mask y>0
//        fused) -> finalize (dgamma/dbeta + normalized sums) -> apply
//        (dx in one pass)
#include "common.h"

namespace {

constexpr int MAXC = 2048;   // LDS scale/shift staging bound (40 KB worst case)

// magic division for the flat-index -> channel decode in the strided-dy
// apply path (see conv.hip for the derivation)
struct FDiv {
  unsigned long long m;
  int s;
};
inline FDiv make_fd(int d) {
  FDiv f;
  int L = 0;
  while ((1LL << L) < d) ++L;
  f.s = 31 + L;
  f.m = ((1ULL << f.s) + d - 1) / d;
  return f;
}
DEVINL unsigned fd(unsigned x, FDiv f) {
  return (unsigned)(((unsigned long long)x * f.m) >> f.s);
}

// grid (ceil(C/64), Z); block 256 = 64 channel lanes x 4 pixel rows
__global__ __launch_bounds__(256)
void bn_stats_kernel(const __bf16* __restrict__ x, float* __restrict__ part,
                     long P, int C, int Z) {
  const int cl = threadIdx.x & 63;
  const int pr = threadIdx.x >> 6;           // 0..3
  const int c = blockIdx.x * 64 + cl;
  const int z = blockIdx.y;
  float sum = 0.f, sq = 0.f;
  if (c < C) {
    // 4 rows in flight per thread: these are independent strided loads
    // and the kernel is latency-bound without the explicit MLP
    const long step = (long)Z * 4;
    long p = (long)z * 4 + pr;
    for (; p + 3 * step < P; p += 4 * step) {
      const float v0 = (float)x[p * C + c];
      const float v1 = (float)x[(p + step) * C + c];
      const float v2 = (float)x[(p + 2 * step) * C + c];
      const float v3 = (float)x[(p + 3 * step) * C + c];
      sum += v0 + v1 + v2 + v3;
      sq += v0 * v0 + v1 * v1 + v2 * v2 + v3 * v3;
    }
    for (; p < P; p += step) {
      const float v = (float)x[p * C + c];
      sum += v;
      sq += v * v;
    }
  }
  __shared__ float ls[4][64], lq[4][64];
  ls[pr][cl] = sum;
  lq[pr][cl] = sq;
  __syncthreads();
  if (pr == 0 && c < C) {
    sum = ls[0][cl] + ls[1][cl] + ls[2][cl] + ls[3][cl];
    sq = lq[0][cl] + lq[1][cl] + lq[2][cl] + lq[3][cl];
    part[((long)z * C + c) * 2] = sum;
    part[((long)z * C + c) * 2 + 1] = sq;
  }
}

// one wave per channel: lanes cover the Z partial slices in parallel
// (a single-workgroup loop over Z was latency-bound at ~28 us)
__global__ __launch_bounds__(64)
void bn_finalize_kernel(const float* __restrict__ part,
                        float* __restrict__ mean,
                        float* __restrict__ invstd,
                        int C, int Z, float inv_count, float eps) {
  const int c = blockIdx.x;
  const int l = threadIdx.x;
  float s = 0.f, q = 0.f;
  for (int z = l; z < Z; z += 64) {
    s += part[((long)z * C + c) * 2];
    q += part[((long)z * C + c) * 2 + 1];
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    s += __shfl_xor(s, off, 64);
    q += __shfl_xor(q, off, 64);
  }
  if (l == 0) {
    const float m = s * inv_count;
    float var = q * inv_count - m * m;
    if (var < 0.f) var = 0.f;
    mean[c] = m;
    invstd[c] = __frsqrt_rn(var + eps);
  }
}

// STRIDED: y is a channel-narrow view of a wider channels-last tensor
// (leading dim ldo > C) — the Inception block writes each branch's BN
// output straight into its slice of the pre-allocated concat buffer,
// eliminating the aten cat copy per block.
template <bool RELU, bool VEC, bool STRIDED>
__global__ __launch_bounds__(256)
void bn_apply_kernel(const __bf16* __restrict__ x,
                     const float* __restrict__ mean,
                     const float* __restrict__ invstd,
                     const __bf16* __restrict__ g, const __bf16* __restrict__ b,
                     __bf16* __restrict__ y, long ldo, FDiv dC,
                     long P, int C) {
  __shared__ float sc[MAXC], sh[MAXC];
  for (int c = threadIdx.x; c < C; c += 256) {
    const float s = (float)g[c] * invstd[c];
    sc[c] = s;
    sh[c] = (float)b[c] - mean[c] * s;
  }
  __syncthreads();
  const long total = P * C;
  const long stride = (long)gridDim.x * 256 * 8;
  for (long i = ((long)blockIdx.x * 256 + threadIdx.x) * 8; i < total;
       i += stride) {
    const long prow = STRIDED ? (long)fd((unsigned)i, dC) : 0;
    const int c0 = STRIDED ? (int)(i - prow * C) : (int)(i % C);
    __bf16* yout = STRIDED ? y + prow * ldo + c0 : y + i;
    if (VEC && i + 8 <= total) {
      bf16x8 v = *(const bf16x8*)&x[i];
      bf16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = (float)v[j] * sc[c0 + j] + sh[c0 + j];
        if (RELU) f = f > 0.f ? f : 0.f;
        o[j] = (__bf16)f;
      }
      *(bf16x8*)yout = o;
    } else {
      int c = c0;
      long jrow = prow;
      for (int j = 0; j < 8 && i + j < total; ++j) {
        float f = (float)x[i + j] * sc[c] + sh[c];
        if (RELU) f = f > 0.f ? f : 0.f;
        if (STRIDED) y[jrow * ldo + c] = (__bf16)f;
        else y[i + j] = (__bf16)f;
        if (++c == C) { c = 0; ++jrow; }
      }
    }
  }
}

// The ReLU mask is recomputed instead of reading the y stream (one
// fewer full activation read), using the SAME expression and bf16
// rounding as the forward apply (y = bf16(x*sc + sh)) so the mask is
// bit-exact with what the forward produced.
template <bool RELU>
__global__ __launch_bounds__(256)
void bn_bwd_stats_kernel(const __bf16* __restrict__ x,
                         const __bf16* __restrict__ dy, long ldy,
                         const __bf16* __restrict__ g,
                         const __bf16* __restrict__ b,
                         const float* __restrict__ mean,
                         const float* __restrict__ invstd,
                         float* __restrict__ part, long P, int C, int Z) {
  const int cl = threadIdx.x & 63;
  const int pr = threadIdx.x >> 6;
  const int c = blockIdx.x * 64 + cl;
  const int z = blockIdx.y;
  float s1 = 0.f, s2 = 0.f;
  if (c < C) {
    const float mu = mean[c], is = invstd[c];
    const float sc = RELU ? (float)g[c] * is : 0.f;
    const float sh = RELU ? (float)b[c] - mu * sc : 0.f;
    const long step = (long)Z * 4;
    long p = (long)z * 4 + pr;
    for (; p + 3 * step < P; p += 4 * step) {
      const long i0 = p * C + c, i1 = i0 + step * C;
      const long i2 = i1 + step * C, i3 = i2 + step * C;
      const long j0 = p * ldy + c, j1 = j0 + step * ldy;
      const long j2 = j1 + step * ldy, j3 = j2 + step * ldy;
      const float x0 = (float)x[i0], x1 = (float)x[i1];
      const float x2 = (float)x[i2], x3 = (float)x[i3];
      float d0 = (float)dy[j0], d1 = (float)dy[j1];
      float d2 = (float)dy[j2], d3 = (float)dy[j3];
      if (RELU && (float)(__bf16)(x0 * sc + sh) <= 0.f) d0 = 0.f;
      if (RELU && (float)(__bf16)(x1 * sc + sh) <= 0.f) d1 = 0.f;
      if (RELU && (float)(__bf16)(x2 * sc + sh) <= 0.f) d2 = 0.f;
      if (RELU && (float)(__bf16)(x3 * sc + sh) <= 0.f) d3 = 0.f;
      s1 += d0 + d1 + d2 + d3;
      s2 += d0 * ((x0 - mu) * is) + d1 * ((x1 - mu) * is) +
            d2 * ((x2 - mu) * is) + d3 * ((x3 - mu) * is);
    }
    for (; p < P; p += step) {
      const float xv = (float)x[p * C + c];
      float d = (float)dy[p * ldy + c];
      const float xh = (xv - mu) * is;
      if (RELU && (float)(__bf16)(xv * sc + sh) <= 0.f) d = 0.f;
      s1 += d;
      s2 += d * xh;
    }
  }
  __shared__ float l1[4][64], l2[4][64];
  l1[pr][cl] = s1;
  l2[pr][cl] = s2;
  __syncthreads();
  if (pr == 0 && c < C) {
    s1 = l1[0][cl] + l1[1][cl] + l1[2][cl] + l1[3][cl];
    s2 = l2[0][cl] + l2[1][cl] + l2[2][cl] + l2[3][cl];
    part[((long)z * C + c) * 2] = s1;
    part[((long)z * C + c) * 2 + 1] = s2;
  }
}

__global__ __launch_bounds__(64)
void bn_bwd_finalize_kernel(const float* __restrict__ part,
                            __bf16* __restrict__ dgamma,
                            __bf16* __restrict__ dbeta,
                            float* __restrict__ s1n,
                            float* __restrict__ s2n,
                            int C, int Z, float inv_count) {
  const int c = blockIdx.x;
  const int l = threadIdx.x;
  float s1 = 0.f, s2 = 0.f;
  for (int z = l; z < Z; z += 64) {
    s1 += part[((long)z * C + c) * 2];
    s2 += part[((long)z * C + c) * 2 + 1];
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    s1 += __shfl_xor(s1, off, 64);
    s2 += __shfl_xor(s2, off, 64);
  }
  if (l == 0) {
    dbeta[c] = (__bf16)s1;
    dgamma[c] = (__bf16)s2;
    s1n[c] = s1 * inv_count;
    s2n[c] = s2 * inv_count;
  }
}

template <bool RELU, bool VEC>
__global__ __launch_bounds__(256)
void bn_bwd_apply_kernel(const __bf16* __restrict__ x,
                         const __bf16* __restrict__ dy, long ldy, FDiv dC,
                         const __bf16* __restrict__ g,
                         const __bf16* __restrict__ b,
                         const float* __restrict__ mean,
                         const float* __restrict__ invstd,
                         const float* __restrict__ s1n,
                         const float* __restrict__ s2n,
                         __bf16* __restrict__ dx, long P, int C) {
  // per-channel constants staged in LDS: dx = gs*(dy_eff - a - xhat*bb),
  // xhat = (x - mu)*is; relu mask recomputed EXACTLY as the forward
  // wrote it: y = bf16(x*sc + sh) (no y read)
  __shared__ float lgs[MAXC], la[MAXC], lbb[MAXC], lmu[MAXC], lis[MAXC];
  __shared__ float lsh[MAXC];
  for (int c = threadIdx.x; c < C; c += 256) {
    const float sc = (float)g[c] * invstd[c];
    lgs[c] = sc;
    la[c] = s1n[c];
    lbb[c] = s2n[c];
    lmu[c] = mean[c];
    lis[c] = invstd[c];
    lsh[c] = (float)b[c] - mean[c] * sc;
  }
  __syncthreads();
  const long total = P * C;
  const long stride = (long)gridDim.x * 256 * 8;
  for (long i = ((long)blockIdx.x * 256 + threadIdx.x) * 8; i < total;
       i += stride) {
    const long prow = fd((unsigned)i, dC);           // i / C (i < 2^31)
    const int c0 = (int)(i - prow * C);
    const long jbase = prow * ldy + c0;
    if (VEC && i + 8 <= total && c0 + 8 <= C) {
      bf16x8 xv = *(const bf16x8*)&x[i];
      bf16x8 dv = *(const bf16x8*)&dy[jbase];
      bf16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int c = c0 + j;
        const float x1 = (float)xv[j];
        float d = (float)dv[j];
        const float xh = (x1 - lmu[c]) * lis[c];
        if (RELU && (float)(__bf16)(x1 * lgs[c] + lsh[c]) <= 0.f) d = 0.f;
        o[j] = (__bf16)(lgs[c] * (d - la[c] - xh * lbb[c]));
      }
      *(bf16x8*)&dx[i] = o;
    } else {
      int c = c0;
      long jrow = prow;
      for (int j = 0; j < 8 && i + j < total; ++j) {
        const float x1 = (float)x[i + j];
        float d = (float)dy[jrow * ldy + c];
        const float xh = (x1 - lmu[c]) * lis[c];
        if (RELU && (float)(__bf16)(x1 * lgs[c] + lsh[c]) <= 0.f) d = 0.f;
        dx[i + j] = (__bf16)(lgs[c] * (d - la[c] - xh * lbb[c]));
        if (++c == C) { c = 0; ++jrow; }
      }
    }
  }
}

// ------------------------------------------------------------ row-wise path
//
// For C % 8 == 0 (every Inception BN layer): thread t owns a FIXED
// 8-channel octet u = t % (C/8) and walks pixel rows. Per-channel
// constants live in REGISTERS (loaded once per thread), the hot loop
// is one b128 load (+1 store) per 8 elements with zero LDS traffic —
// the flat-walk kernels below gathered 6 per-channel constants from
// LDS per element and the PMC showed bank-conflict counts 10x the LDS
// instruction count (profiles/r01_inception_pmc.txt), capping them at
// 1.7-3 TB/s on an 6.3 TB/s chip.
// Stats partials are written [c][z] (channel-major) so the finalize
// lanes read CONSECUTIVE z — coalesced — instead of striding C.

// grid (Z); block 256 = (C/8) channel octets x rpb rows
__global__ __launch_bounds__(256)
void bn_stats_rw_kernel(const __bf16* __restrict__ x,
                        float* __restrict__ part,
                        long P, int C, int Z) {
  const int tpr = C >> 3;
  const int u = (int)(threadIdx.x % tpr);
  const int rl = (int)(threadIdx.x / tpr);
  const int rpb = 256 / tpr < 1 ? 1 : 256 / tpr;
  const int c0 = u * 8;
  const int z = blockIdx.x;
  float sum[8] = {}, sq[8] = {};
  if (rl < rpb) {
    const long rstep = (long)Z * rpb;
    long r = (long)z * rpb + rl;
    // 4 rows of loads in flight: the single-load loop was
    // latency-bound (one dependent accumulate per HBM round trip)
    for (; r + 3 * rstep < P; r += 4 * rstep) {
      const bf16x8 v0 = *(const bf16x8*)&x[r * C + c0];
      const bf16x8 v1 = *(const bf16x8*)&x[(r + rstep) * C + c0];
      const bf16x8 v2 = *(const bf16x8*)&x[(r + 2 * rstep) * C + c0];
      const bf16x8 v3 = *(const bf16x8*)&x[(r + 3 * rstep) * C + c0];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float f0 = (float)v0[j], f1 = (float)v1[j];
        const float f2 = (float)v2[j], f3 = (float)v3[j];
        sum[j] += (f0 + f1) + (f2 + f3);
        sq[j] += (f0 * f0 + f1 * f1) + (f2 * f2 + f3 * f3);
      }
    }
    for (; r < P; r += rstep) {
      const bf16x8 v = *(const bf16x8*)&x[r * C + c0];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float f = (float)v[j];
        sum[j] += f;
        sq[j] += f * f;
      }
    }
  }
  // cross-row tree reduce (log2(rpb) rounds, all rows participating —
  // a serial per-thread loop here left tpr of 256 lanes active and
  // dominated the big-layer kernels)
  __shared__ float red[256 * 8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    red[threadIdx.x * 8 + j] = sum[j];
  }
  __syncthreads();
  for (int cur = rpb; cur > 1;) {
    const int half = cur >> 1;
    if (rl < cur - half) {   // fold rows [half, cur) onto [0, cur-half)
#pragma unroll
      for (int j = 0; j < 8; ++j)
        red[threadIdx.x * 8 + j] += red[((rl + half) * tpr + u) * 8 + j];
    }
    __syncthreads();
    cur -= half;
  }
  if (rl == 0) {
#pragma unroll
    for (int j = 0; j < 8; ++j) sum[j] = red[u * 8 + j];
  }
  __syncthreads();
#pragma unroll
  for (int j = 0; j < 8; ++j) red[threadIdx.x * 8 + j] = sq[j];
  __syncthreads();
  for (int cur = rpb; cur > 1;) {
    const int half = cur >> 1;
    if (rl < cur - half) {
#pragma unroll
      for (int j = 0; j < 8; ++j)
        red[threadIdx.x * 8 + j] += red[((rl + half) * tpr + u) * 8 + j];
    }
    __syncthreads();
    cur -= half;
  }
  if (rl == 0) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      // channel-major partials: [c][z]{sum, sq}
      part[((long)(c0 + j) * Z + z) * 2] = sum[j];
      part[((long)(c0 + j) * Z + z) * 2 + 1] = red[u * 8 + j];
    }
  }
}

// block 256 = one channel; lanes stride z (coalesced in the [c][z]
// layout — Z can be 2048, so a whole block per channel keeps enough
// loads in flight)
__global__ __launch_bounds__(256)
void bn_finalize_rw_kernel(const float* __restrict__ part,
                           float* __restrict__ mean,
                           float* __restrict__ invstd,
                           int C, int Z, float inv_count, float eps) {
  const int c = blockIdx.x;
  const int t = threadIdx.x;
  float s = 0.f, q = 0.f;
  for (int z = t; z < Z; z += 256) {
    s += part[((long)c * Z + z) * 2];
    q += part[((long)c * Z + z) * 2 + 1];
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    s += __shfl_xor(s, off, 64);
    q += __shfl_xor(q, off, 64);
  }
  __shared__ float ls[4], lq[4];
  if ((t & 63) == 0) {
    ls[t >> 6] = s;
    lq[t >> 6] = q;
  }
  __syncthreads();
  if (t == 0) {
    s = ls[0] + ls[1] + ls[2] + ls[3];
    q = lq[0] + lq[1] + lq[2] + lq[3];
    const float m = s * inv_count;
    float var = q * inv_count - m * m;
    if (var < 0.f) var = 0.f;
    mean[c] = m;
    invstd[c] = __frsqrt_rn(var + eps);
  }
}

template <bool RELU>
__global__ __launch_bounds__(256)
void bn_apply_rw_kernel(const __bf16* __restrict__ x,
                        const float* __restrict__ mean,
                        const float* __restrict__ invstd,
                        const __bf16* __restrict__ g,
                        const __bf16* __restrict__ b,
                        __bf16* __restrict__ y, long ldo, long P, int C) {
  const int tpr = C >> 3;
  const int u = (int)(threadIdx.x % tpr);
  const int rl = (int)(threadIdx.x / tpr);
  const int rpb = 256 / tpr < 1 ? 1 : 256 / tpr;
  if (rl >= rpb) return;
  const int c0 = u * 8;
  float sc[8], sh[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const float s = (float)g[c0 + j] * invstd[c0 + j];
    sc[j] = s;
    sh[j] = (float)b[c0 + j] - mean[c0 + j] * s;
  }
  const long rstep = (long)gridDim.x * rpb;
  long r = (long)blockIdx.x * rpb + rl;
  for (; r + rstep < P; r += 2 * rstep) {   // 2 loads in flight
    const bf16x8 v0 = *(const bf16x8*)&x[r * C + c0];
    const bf16x8 v1 = *(const bf16x8*)&x[(r + rstep) * C + c0];
    bf16x8 o0, o1;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f0 = (float)v0[j] * sc[j] + sh[j];
      float f1 = (float)v1[j] * sc[j] + sh[j];
      if (RELU) {
        f0 = f0 > 0.f ? f0 : 0.f;
        f1 = f1 > 0.f ? f1 : 0.f;
      }
      o0[j] = (__bf16)f0;
      o1[j] = (__bf16)f1;
    }
    *(bf16x8*)&y[r * ldo + c0] = o0;
    *(bf16x8*)&y[(r + rstep) * ldo + c0] = o1;
  }
  for (; r < P; r += rstep) {
    const bf16x8 v = *(const bf16x8*)&x[r * C + c0];
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = (float)v[j] * sc[j] + sh[j];
      if (RELU) f = f > 0.f ? f : 0.f;
      o[j] = (__bf16)f;
    }
    *(bf16x8*)&y[r * ldo + c0] = o;
  }
}

template <bool RELU>
__global__ __launch_bounds__(256)
void bn_bwd_stats_rw_kernel(const __bf16* __restrict__ x,
                            const __bf16* __restrict__ dy, long ldy,
                            const __bf16* __restrict__ g,
                            const __bf16* __restrict__ b,
                            const float* __restrict__ mean,
                            const float* __restrict__ invstd,
                            float* __restrict__ part, long P, int C, int Z) {
  const int tpr = C >> 3;
  const int u = (int)(threadIdx.x % tpr);
  const int rl = (int)(threadIdx.x / tpr);
  const int rpb = 256 / tpr < 1 ? 1 : 256 / tpr;
  const int c0 = u * 8;
  const int z = blockIdx.x;
  float s1[8] = {}, s2[8] = {};
  if (rl < rpb) {
    float mu[8], is[8], sc[8], sh[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      mu[j] = mean[c0 + j];
      is[j] = invstd[c0 + j];
      const float s = RELU ? (float)g[c0 + j] * is[j] : 0.f;
      sc[j] = s;
      sh[j] = RELU ? (float)b[c0 + j] - mu[j] * s : 0.f;
    }
    const long rstep = (long)Z * rpb;
    long r = (long)z * rpb + rl;
    // 2 rows x (x, dy) = 4 loads in flight (see bn_stats_rw_kernel)
    for (; r + rstep < P; r += 2 * rstep) {
      const bf16x8 xv0 = *(const bf16x8*)&x[r * C + c0];
      const bf16x8 dv0 = *(const bf16x8*)&dy[r * ldy + c0];
      const bf16x8 xv1 = *(const bf16x8*)&x[(r + rstep) * C + c0];
      const bf16x8 dv1 = *(const bf16x8*)&dy[(r + rstep) * ldy + c0];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float x0 = (float)xv0[j], x1 = (float)xv1[j];
        float d0 = (float)dv0[j], d1 = (float)dv1[j];
        // relu mask recomputed EXACTLY as the forward wrote it
        if (RELU && (float)(__bf16)(x0 * sc[j] + sh[j]) <= 0.f) d0 = 0.f;
        if (RELU && (float)(__bf16)(x1 * sc[j] + sh[j]) <= 0.f) d1 = 0.f;
        s1[j] += d0 + d1;
        s2[j] += d0 * ((x0 - mu[j]) * is[j]) + d1 * ((x1 - mu[j]) * is[j]);
      }
    }
    for (; r < P; r += rstep) {
      const bf16x8 xv = *(const bf16x8*)&x[r * C + c0];
      const bf16x8 dv = *(const bf16x8*)&dy[r * ldy + c0];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float xf = (float)xv[j];
        float d = (float)dv[j];
        if (RELU && (float)(__bf16)(xf * sc[j] + sh[j]) <= 0.f) d = 0.f;
        s1[j] += d;
        s2[j] += d * ((xf - mu[j]) * is[j]);
      }
    }
  }
  __shared__ float red[256 * 8];
#pragma unroll
  for (int j = 0; j < 8; ++j) red[threadIdx.x * 8 + j] = s1[j];
  __syncthreads();
  for (int cur = rpb; cur > 1;) {
    const int half = cur >> 1;
    if (rl < cur - half) {   // tree reduce (see bn_stats_rw_kernel)
#pragma unroll
      for (int j = 0; j < 8; ++j)
        red[threadIdx.x * 8 + j] += red[((rl + half) * tpr + u) * 8 + j];
    }
    __syncthreads();
    cur -= half;
  }
  if (rl == 0) {
#pragma unroll
    for (int j = 0; j < 8; ++j) s1[j] = red[u * 8 + j];
  }
  __syncthreads();
#pragma unroll
  for (int j = 0; j < 8; ++j) red[threadIdx.x * 8 + j] = s2[j];
  __syncthreads();
  for (int cur = rpb; cur > 1;) {
    const int half = cur >> 1;
    if (rl < cur - half) {
#pragma unroll
      for (int j = 0; j < 8; ++j)
        red[threadIdx.x * 8 + j] += red[((rl + half) * tpr + u) * 8 + j];
    }
    __syncthreads();
    cur -= half;
  }
  if (rl == 0) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      part[((long)(c0 + j) * Z + z) * 2] = s1[j];
      part[((long)(c0 + j) * Z + z) * 2 + 1] = red[u * 8 + j];
    }
  }
}

__global__ __launch_bounds__(256)
void bn_bwd_finalize_rw_kernel(const float* __restrict__ part,
                               __bf16* __restrict__ dgamma,
                               __bf16* __restrict__ dbeta,
                               float* __restrict__ s1n,
                               float* __restrict__ s2n,
                               int C, int Z, float inv_count) {
  const int c = blockIdx.x;
  const int t = threadIdx.x;
  float s1 = 0.f, s2 = 0.f;
  for (int z = t; z < Z; z += 256) {
    s1 += part[((long)c * Z + z) * 2];
    s2 += part[((long)c * Z + z) * 2 + 1];
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    s1 += __shfl_xor(s1, off, 64);
    s2 += __shfl_xor(s2, off, 64);
  }
  __shared__ float l1[4], l2[4];
  if ((t & 63) == 0) {
    l1[t >> 6] = s1;
    l2[t >> 6] = s2;
  }
  __syncthreads();
  if (t == 0) {
    s1 = l1[0] + l1[1] + l1[2] + l1[3];
    s2 = l2[0] + l2[1] + l2[2] + l2[3];
    dbeta[c] = (__bf16)s1;
    dgamma[c] = (__bf16)s2;
    s1n[c] = s1 * inv_count;
    s2n[c] = s2 * inv_count;
  }
}

template <bool RELU>
__global__ __launch_bounds__(256)
void bn_bwd_apply_rw_kernel(const __bf16* __restrict__ x,
                            const __bf16* __restrict__ dy, long ldy,
                            const __bf16* __restrict__ g,
                            const __bf16* __restrict__ b,
                            const float* __restrict__ mean,
                            const float* __restrict__ invstd,
                            const float* __restrict__ s1n,
                            const float* __restrict__ s2n,
                            __bf16* __restrict__ dx, long P, int C) {
  const int tpr = C >> 3;
  const int u = (int)(threadIdx.x % tpr);
  const int rl = (int)(threadIdx.x / tpr);
  const int rpb = 256 / tpr < 1 ? 1 : 256 / tpr;
  if (rl >= rpb) return;
  const int c0 = u * 8;
  float gs[8], a[8], bb[8], mu[8], is[8], sh[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const int c = c0 + j;
    is[j] = invstd[c];
    mu[j] = mean[c];
    const float s = (float)g[c] * is[j];
    gs[j] = s;
    a[j] = s1n[c];
    bb[j] = s2n[c];
    sh[j] = (float)b[c] - mu[j] * s;
  }
  const long rstep = (long)gridDim.x * rpb;
  long r = (long)blockIdx.x * rpb + rl;
  for (; r + rstep < P; r += 2 * rstep) {   // 4 loads in flight
    const bf16x8 xv0 = *(const bf16x8*)&x[r * C + c0];
    const bf16x8 dv0 = *(const bf16x8*)&dy[r * ldy + c0];
    const bf16x8 xv1 = *(const bf16x8*)&x[(r + rstep) * C + c0];
    const bf16x8 dv1 = *(const bf16x8*)&dy[(r + rstep) * ldy + c0];
    bf16x8 o0, o1;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float x0 = (float)xv0[j], x1 = (float)xv1[j];
      float d0 = (float)dv0[j], d1 = (float)dv1[j];
      const float xh0 = (x0 - mu[j]) * is[j];
      const float xh1 = (x1 - mu[j]) * is[j];
      if (RELU && (float)(__bf16)(x0 * gs[j] + sh[j]) <= 0.f) d0 = 0.f;
      if (RELU && (float)(__bf16)(x1 * gs[j] + sh[j]) <= 0.f) d1 = 0.f;
      o0[j] = (__bf16)(gs[j] * (d0 - a[j] - xh0 * bb[j]));
      o1[j] = (__bf16)(gs[j] * (d1 - a[j] - xh1 * bb[j]));
    }
    *(bf16x8*)&dx[r * C + c0] = o0;
    *(bf16x8*)&dx[(r + rstep) * C + c0] = o1;
  }
  for (; r < P; r += rstep) {
    const bf16x8 xv = *(const bf16x8*)&x[r * C + c0];
    const bf16x8 dv = *(const bf16x8*)&dy[r * ldy + c0];
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float xf = (float)xv[j];
      float d = (float)dv[j];
      const float xh = (xf - mu[j]) * is[j];
      if (RELU && (float)(__bf16)(xf * gs[j] + sh[j]) <= 0.f) d = 0.f;
      o[j] = (__bf16)(gs[j] * (d - a[j] - xh * bb[j]));
    }
    *(bf16x8*)&dx[r * C + c0] = o;
  }
}

inline bool rowwise_ok(int C) { return (C & 7) == 0 && C <= MAXC; }

// ------------------------------------------------------------ grouped path
//
// One launch per phase for ALL terminal BNs of an Inception block
// (A/C/E blocks end in 4-6 independent conv->BN branches writing one
// concat buffer): per-branch x/dy/param pointers ride a kernel arg
// struct and grid.y (or a channel-block table) selects the branch.
// BN is per-channel, so the grouped result is IDENTICAL to separate
// per-branch BNs — only the launch count changes (6 kernels per
// BRANCH chain -> 6 per BLOCK; the per-kernel execution floor at
// these layer sizes made launch count the dominant BN cost).
constexpr int BNG_MAX = 8;

struct BNGroupArgs {
  const __bf16* x[BNG_MAX];    // per-branch pre-BN conv output [P, C_b]
  const __bf16* dy[BNG_MAX];   // bwd: per-branch dy base
  long dyld[BNG_MAX];          // bwd: per-branch dy row stride
  __bf16* dx[BNG_MAX];         // bwd: per-branch dx out [P, C_b]
  __bf16* y[BNG_MAX];          // fwd: per-branch out base
  long yld[BNG_MAX];           // fwd: per-branch out row stride
  const __bf16* g[BNG_MAX];
  const __bf16* b[BNG_MAX];
  int C[BNG_MAX];
  int coff[BNG_MAX];           // channel offset within the group
  int cb0[BNG_MAX];            // first 64-wide channel block index
  FDiv dC[BNG_MAX];            // magic divide by C (flat-walk decode)
  int n;
};

DEVINL int bng_branch(const BNGroupArgs& a, int cb) {
  int br = 0;
  for (int i = 1; i < BNG_MAX; ++i)
    if (i < a.n && cb >= a.cb0[i]) br = i;
  return br;
}

// grid (cbtot, Z); column-walk stats (the grouped layers are the
// small-P blocks where this granularity wins — see rowwise_stats)
__global__ __launch_bounds__(256)
void bng_stats_kernel(BNGroupArgs a, float* __restrict__ part,
                      long P, int Ctot, int Z) {
  const int cl = threadIdx.x & 63;
  const int pr = threadIdx.x >> 6;
  const int br = bng_branch(a, blockIdx.x);
  const __bf16* x = a.x[br];
  const int C = a.C[br];
  const int c = (blockIdx.x - a.cb0[br]) * 64 + cl;
  const int z = blockIdx.y;
  float sum = 0.f, sq = 0.f;
  if (c < C) {
    const long step = (long)Z * 4;
    long p = (long)z * 4 + pr;
    for (; p + 3 * step < P; p += 4 * step) {
      const float v0 = (float)x[p * C + c];
      const float v1 = (float)x[(p + step) * C + c];
      const float v2 = (float)x[(p + 2 * step) * C + c];
      const float v3 = (float)x[(p + 3 * step) * C + c];
      sum += v0 + v1 + v2 + v3;
      sq += v0 * v0 + v1 * v1 + v2 * v2 + v3 * v3;
    }
    for (; p < P; p += step) {
      const float v = (float)x[p * C + c];
      sum += v;
      sq += v * v;
    }
  }
  __shared__ float ls[4][64], lq[4][64];
  ls[pr][cl] = sum;
  lq[pr][cl] = sq;
  __syncthreads();
  if (pr == 0 && c < C) {
    sum = ls[0][cl] + ls[1][cl] + ls[2][cl] + ls[3][cl];
    sq = lq[0][cl] + lq[1][cl] + lq[2][cl] + lq[3][cl];
    const int gc = a.coff[br] + c;
    part[((long)z * Ctot + gc) * 2] = sum;
    part[((long)z * Ctot + gc) * 2 + 1] = sq;
  }
}

// grid (ew blocks, n): per-branch flat walk, strided store into the
// concat slice (out + coff), per-channel constants staged in LDS
template <bool RELU>
__global__ __launch_bounds__(256)
void bng_apply_kernel(BNGroupArgs a, const float* __restrict__ mean,
                      const float* __restrict__ invstd, long P) {
  const int br = blockIdx.y;
  const __bf16* x = a.x[br];
  const int C = a.C[br];
  const int coff = a.coff[br];
  const __bf16* g = a.g[br];
  const __bf16* bb = a.b[br];
  __bf16* y = a.y[br];
  const long ldo = a.yld[br];
  __shared__ float sc[512], sh[512];
  for (int c = threadIdx.x; c < C; c += 256) {
    const float s = (float)g[c] * invstd[coff + c];
    sc[c] = s;
    sh[c] = (float)bb[c] - mean[coff + c] * s;
  }
  __syncthreads();
  const long total = P * C;
  const long stride = (long)gridDim.x * 256 * 8;
  const FDiv dC = a.dC[br];
  for (long i = ((long)blockIdx.x * 256 + threadIdx.x) * 8; i < total;
       i += stride) {
    const long prow = fd((unsigned)i, dC);  // C%8==0: chunk stays in-row
    const int c0 = (int)(i - prow * C);
    bf16x8 v = *(const bf16x8*)&x[i];
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = (float)v[j] * sc[c0 + j] + sh[c0 + j];
      if (RELU) f = f > 0.f ? f : 0.f;
      o[j] = (__bf16)f;
    }
    *(bf16x8*)&y[prow * ldo + c0] = o;
  }
}

template <bool RELU>
__global__ __launch_bounds__(256)
void bng_bwd_stats_kernel(BNGroupArgs a,
                          const float* __restrict__ mean,
                          const float* __restrict__ invstd,
                          float* __restrict__ part, long P, int Ctot,
                          int Z) {
  const int cl = threadIdx.x & 63;
  const int pr = threadIdx.x >> 6;
  const int br = bng_branch(a, blockIdx.x);
  const __bf16* x = a.x[br];
  const __bf16* dy = a.dy[br];
  const long ldy = a.dyld[br];
  const int C = a.C[br];
  const int c = (blockIdx.x - a.cb0[br]) * 64 + cl;
  const int gc = a.coff[br] + c;
  const int z = blockIdx.y;
  float s1 = 0.f, s2 = 0.f;
  if (c < C) {
    const float mu = mean[gc], is = invstd[gc];
    const float sc = RELU ? (float)a.g[br][c] * is : 0.f;
    const float sh = RELU ? (float)a.b[br][c] - mu * sc : 0.f;
    const long step = (long)Z * 4;
    long p = (long)z * 4 + pr;
    for (; p < P; p += step) {
      const float xv = (float)x[p * C + c];
      float d = (float)dy[p * ldy + c];
      const float xh = (xv - mu) * is;
      if (RELU && (float)(__bf16)(xv * sc + sh) <= 0.f) d = 0.f;
      s1 += d;
      s2 += d * xh;
    }
  }
  __shared__ float l1[4][64], l2[4][64];
  l1[pr][cl] = s1;
  l2[pr][cl] = s2;
  __syncthreads();
  if (pr == 0 && c < C) {
    s1 = l1[0][cl] + l1[1][cl] + l1[2][cl] + l1[3][cl];
    s2 = l2[0][cl] + l2[1][cl] + l2[2][cl] + l2[3][cl];
    part[((long)z * Ctot + gc) * 2] = s1;
    part[((long)z * Ctot + gc) * 2 + 1] = s2;
  }
}

template <bool RELU>
__global__ __launch_bounds__(256)
void bng_bwd_apply_kernel(BNGroupArgs a,
                          const float* __restrict__ mean,
                          const float* __restrict__ invstd,
                          const float* __restrict__ s1n,
                          const float* __restrict__ s2n, long P) {
  const int br = blockIdx.y;
  const __bf16* x = a.x[br];
  const __bf16* dy = a.dy[br];
  const long ldy = a.dyld[br];
  __bf16* dx = a.dx[br];
  const int C = a.C[br];
  const int coff = a.coff[br];
  __shared__ float lgs[512], la[512], lbb[512], lmu[512], lis[512];
  __shared__ float lsh[512];
  for (int c = threadIdx.x; c < C; c += 256) {
    const int gc = coff + c;
    const float s = (float)a.g[br][c] * invstd[gc];
    lgs[c] = s;
    la[c] = s1n[gc];
    lbb[c] = s2n[gc];
    lmu[c] = mean[gc];
    lis[c] = invstd[gc];
    lsh[c] = (float)a.b[br][c] - mean[gc] * s;
  }
  __syncthreads();
  const long total = P * C;
  const long stride = (long)gridDim.x * 256 * 8;
  const FDiv dC = a.dC[br];
  for (long i = ((long)blockIdx.x * 256 + threadIdx.x) * 8; i < total;
       i += stride) {
    const long prow = fd((unsigned)i, dC);
    const int c0 = (int)(i - prow * C);
    const bf16x8 xv = *(const bf16x8*)&x[i];
    const bf16x8 dv = *(const bf16x8*)&dy[prow * ldy + c0];
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int c = c0 + j;
      const float x1 = (float)xv[j];
      float d = (float)dv[j];
      const float xh = (x1 - lmu[c]) * lis[c];
      if (RELU && (float)(__bf16)(x1 * lgs[c] + lsh[c]) <= 0.f) d = 0.f;
      o[j] = (__bf16)(lgs[c] * (d - la[c] - xh * lbb[c]));
    }
    *(bf16x8*)&dx[i] = o;
  }
}

inline unsigned rw_grid(long P, int C) {
  const int rpb = 256 / (C >> 3) < 1 ? 1 : 256 / (C >> 3);
  long wgs = (P + rpb - 1) / rpb;
  if (wgs > 4096) wgs = 4096;
  if (wgs < 1) wgs = 1;
  return (unsigned)wgs;
}

// Row-wise stats only where the 8-channels-per-thread granularity
// still yields enough BLOCKS to cover the chip (>=512 at 2 passes per
// thread): small-P layers have 8x more parallelism at the old
// one-channel-per-thread granularity and measured FASTER there (the
// first rowwise cut regressed mid-size layers 2x on exactly this).
inline bool rowwise_stats(long P, int C) {
  if (!rowwise_ok(C)) return false;
  const int t = 256 / (C >> 3);
  const int rpb = t < 1 ? 1 : t;
  return P / ((long)rpb * 2) >= 512;
}

inline int stats_slices(long P, int C) {
  if (rowwise_stats(P, C)) {
    // row-wise stats: grid = Z blocks, each covering all C and rpb
    // rows per pass; ~2 passes per thread (2 loads in flight), more
    // via the 4-deep unroll when P allows
    const int t = 256 / (C >> 3);
    const int rpb = t < 1 ? 1 : t;
    long z = (P + 2L * rpb - 1) / (2L * rpb);
    if (z > 2048) z = 2048;
    if (z < 1) z = 1;
    return (int)z;
  }
  // target >=2048 workgroups across the (C/64) x Z grid, each slice
  // covering >=8 pixel rounds of 4 rows. The cap matters: narrow
  // layers (C=64 -> one channel column) need Z ~ 2048 to cover the
  // 256-CU chip; the old 256 cap left them 1 workgroup/CU and the PMC
  // wait counters dominated everything else in the stats kernels.
  long cb = (C + 63) / 64;
  long want = (2048 + cb - 1) / cb;
  long per = P / (4 * 8);
  long z = want < per ? want : per;
  if (z < 1) z = 1;
  if (z > 2048) z = 2048;
  return (int)z;
}

inline unsigned ew_grid(long total) {
  long wgs = (total / 8 + 255) / 256;
  if (wgs > 8192) wgs = 8192;
  if (wgs < 1) wgs = 1;
  return (unsigned)wgs;
}

}  // namespace

void launch_bn_fwd(const bf16_t* x, const bf16_t* g, const bf16_t* b,
                   bf16_t* y, long ldo, float* mean, float* invstd,
                   float* part, long P, int C, int Z, float eps, bool relu,
                   hipStream_t stream) {
  if (rowwise_stats(P, C)) {
    hipLaunchKernelGGL(bn_stats_rw_kernel, dim3(Z), dim3(256), 0, stream,
                       (const __bf16*)x, part, P, C, Z);
    hipLaunchKernelGGL(bn_finalize_rw_kernel, dim3(C),
                       dim3(256), 0, stream, part, mean, invstd, C, Z,
                       1.f / (float)P, eps);
  } else {
    dim3 sg(ceil_div(C, 64), Z), sb(256);
    hipLaunchKernelGGL(bn_stats_kernel, sg, sb, 0, stream, (const __bf16*)x,
                       part, P, C, Z);
    hipLaunchKernelGGL(bn_finalize_kernel, dim3(C), dim3(64),
                       0, stream, part, mean, invstd, C, Z, 1.f / (float)P,
                       eps);
  }
  if (rowwise_ok(C)) {
    dim3 ag(rw_grid(P, C)), ab(256);
    if (relu)
      hipLaunchKernelGGL((bn_apply_rw_kernel<true>), ag, ab, 0, stream,
                         (const __bf16*)x, mean, invstd, (const __bf16*)g,
                         (const __bf16*)b, (__bf16*)y, ldo, P, C);
    else
      hipLaunchKernelGGL((bn_apply_rw_kernel<false>), ag, ab, 0, stream,
                         (const __bf16*)x, mean, invstd, (const __bf16*)g,
                         (const __bf16*)b, (__bf16*)y, ldo, P, C);
    return;
  }
  dim3 ag(ew_grid(P * (long)C)), ab(256);
  const bool vec = (C & 7) == 0;
  const bool strided = ldo != C;
  const FDiv dC = make_fd(C);
#define APPLY(RELUv, VECv, STRv)                                            \
  hipLaunchKernelGGL((bn_apply_kernel<RELUv, VECv, STRv>), ag, ab, 0,       \
                     stream, (const __bf16*)x, mean, invstd,                 \
                     (const __bf16*)g, (const __bf16*)b, (__bf16*)y, ldo,    \
                     dC, P, C)
#define APPLY2(RELUv, VECv)                                                 \
  do { if (strided) APPLY(RELUv, VECv, true);                               \
       else APPLY(RELUv, VECv, false); } while (0)
  if (relu) { if (vec) APPLY2(true, true); else APPLY2(true, false); }
  else      { if (vec) APPLY2(false, true); else APPLY2(false, false); }
#undef APPLY2
#undef APPLY
}

void launch_bn_bwd(const bf16_t* x, const bf16_t* dy, long ldy,
                   const bf16_t* g, const bf16_t* b, const float* mean,
                   const float* invstd, bf16_t* dx, bf16_t* dgamma,
                   bf16_t* dbeta, float* part, float* s1n, float* s2n,
                   long P, int C, int Z, bool relu, hipStream_t stream) {
  if (rowwise_stats(P, C)) {
#define RWS(RELUv)                                                          \
    hipLaunchKernelGGL((bn_bwd_stats_rw_kernel<RELUv>), dim3(Z), dim3(256), \
                       0, stream, (const __bf16*)x, (const __bf16*)dy, ldy, \
                       (const __bf16*)g, (const __bf16*)b, mean, invstd,    \
                       part, P, C, Z)
    if (relu) RWS(true); else RWS(false);
#undef RWS
    hipLaunchKernelGGL(bn_bwd_finalize_rw_kernel, dim3(C),
                       dim3(256), 0, stream, part, (__bf16*)dgamma,
                       (__bf16*)dbeta, s1n, s2n, C, Z, 1.f / (float)P);
  } else {
    dim3 sg(ceil_div(C, 64), Z), sb(256);
    if (relu)
      hipLaunchKernelGGL((bn_bwd_stats_kernel<true>), sg, sb, 0, stream,
                         (const __bf16*)x, (const __bf16*)dy, ldy,
                         (const __bf16*)g, (const __bf16*)b, mean, invstd,
                         part, P, C, Z);
    else
      hipLaunchKernelGGL((bn_bwd_stats_kernel<false>), sg, sb, 0, stream,
                         (const __bf16*)x, (const __bf16*)dy, ldy,
                         (const __bf16*)g, (const __bf16*)b, mean, invstd,
                         part, P, C, Z);
    hipLaunchKernelGGL(bn_bwd_finalize_kernel, dim3(C), dim3(64), 0, stream,
                       part, (__bf16*)dgamma, (__bf16*)dbeta, s1n, s2n, C, Z,
                       1.f / (float)P);
  }
  if (rowwise_ok(C)) {
    dim3 ag(rw_grid(P, C)), ab(256);
#define RWA(RELUv)                                                          \
    hipLaunchKernelGGL((bn_bwd_apply_rw_kernel<RELUv>), ag, ab, 0, stream,  \
                       (const __bf16*)x, (const __bf16*)dy, ldy,            \
                       (const __bf16*)g, (const __bf16*)b, mean, invstd,    \
                       s1n, s2n, (__bf16*)dx, P, C)
    if (relu) RWA(true); else RWA(false);
#undef RWA
    return;
  }
  dim3 ag(ew_grid(P * (long)C)), ab(256);
  const bool vec = (C & 7) == 0;
  const FDiv dC = make_fd(C);
#define APPLY(RELUv, VECv)                                                  \
  hipLaunchKernelGGL((bn_bwd_apply_kernel<RELUv, VECv>), ag, ab, 0, stream, \
                     (const __bf16*)x, (const __bf16*)dy, ldy, dC,           \
                     (const __bf16*)g, (const __bf16*)b, mean, invstd, s1n,  \
                     s2n, (__bf16*)dx, P, C)
  if (relu) { if (vec) APPLY(true, true); else APPLY(true, false); }
  else      { if (vec) APPLY(false, true); else APPLY(false, false); }
#undef APPLY
}

int bn_stats_slices(long P, int C) { return stats_slices(P, C); }
int bn_max_channels() { return MAXC; }

// -------------------------------------------------------- grouped launchers

int bn_group_slices(long P, const int* Cs, int n) {
  long cbtot = 0;
  for (int i = 0; i < n; ++i) cbtot += ceil_div(Cs[i], 64);
  long want = (2048 + cbtot - 1) / cbtot;
  long per = P / (4 * 8);
  long z = want < per ? want : per;
  if (z < 1) z = 1;
  if (z > 2048) z = 2048;
  return (int)z;
}

namespace {

BNGroupArgs fill_group_args(const bf16_t* const* xs, const bf16_t* const* gs,
                            const bf16_t* const* bs, const int* Cs, int n,
                            const bf16_t* const* dys, const long* dylds,
                            bf16_t* const* dxs,
                            bf16_t* const* ys, const long* ylds,
                            int* cbtot_out, int* ctot_out) {
  BNGroupArgs a = {};
  a.n = n;
  int coff = 0, cb = 0;
  for (int i = 0; i < n; ++i) {
    a.x[i] = (const __bf16*)xs[i];
    a.g[i] = (const __bf16*)gs[i];
    a.b[i] = (const __bf16*)bs[i];
    a.dy[i] = dys ? (const __bf16*)dys[i] : nullptr;
    a.dyld[i] = dylds ? dylds[i] : 0;
    a.dx[i] = dxs ? (__bf16*)dxs[i] : nullptr;
    a.y[i] = ys ? (__bf16*)ys[i] : nullptr;
    a.yld[i] = ylds ? ylds[i] : 0;
    a.C[i] = Cs[i];
    a.coff[i] = coff;
    a.cb0[i] = cb;
    a.dC[i] = make_fd(Cs[i]);
    coff += Cs[i];
    cb += ceil_div(Cs[i], 64);
  }
  *cbtot_out = cb;
  *ctot_out = coff;
  return a;
}

inline unsigned bng_ew_grid(long P, int cmax) {
  long wgs = ((P * cmax) / 8 + 255) / 256;
  if (wgs > 2048) wgs = 2048;
  if (wgs < 1) wgs = 1;
  return (unsigned)wgs;
}

}  // namespace

void launch_bn_group_fwd(const bf16_t* const* xs, const bf16_t* const* gs,
                         const bf16_t* const* bs, const int* Cs, int n,
                         bf16_t* const* youts, const long* ylds,
                         float* mean, float* invstd,
                         float* part, long P, int Z, float eps, bool relu,
                         hipStream_t stream) {
  int cbtot, ctot;
  BNGroupArgs a = fill_group_args(xs, gs, bs, Cs, n, nullptr, nullptr,
                                  nullptr, youts, ylds, &cbtot, &ctot);
  hipLaunchKernelGGL(bng_stats_kernel, dim3(cbtot, Z), dim3(256), 0, stream,
                     a, part, P, ctot, Z);
  hipLaunchKernelGGL(bn_finalize_kernel, dim3(ctot), dim3(64), 0, stream,
                     part, mean, invstd, ctot, Z, 1.f / (float)P, eps);
  int cmax = 0;
  for (int i = 0; i < n; ++i) cmax = Cs[i] > cmax ? Cs[i] : cmax;
  dim3 ag(bng_ew_grid(P, cmax), n), ab(256);
  if (relu)
    hipLaunchKernelGGL((bng_apply_kernel<true>), ag, ab, 0, stream, a, mean,
                       invstd, P);
  else
    hipLaunchKernelGGL((bng_apply_kernel<false>), ag, ab, 0, stream, a, mean,
                       invstd, P);
}

void launch_bn_group_bwd(const bf16_t* const* xs, const bf16_t* const* dys,
                         const long* dylds, const bf16_t* const* gs,
                         const bf16_t* const* bs, bf16_t* const* dxs,
                         const int* Cs, int n, const float* mean,
                         const float* invstd, bf16_t* dgamma, bf16_t* dbeta,
                         float* s1n, float* s2n, float* part, long P, int Z,
                         bool relu, hipStream_t stream) {
  int cbtot, ctot;
  BNGroupArgs a = fill_group_args(xs, gs, bs, Cs, n, dys, dylds, dxs,
                                  nullptr, nullptr, &cbtot, &ctot);
  if (relu)
    hipLaunchKernelGGL((bng_bwd_stats_kernel<true>), dim3(cbtot, Z),
                       dim3(256), 0, stream, a, mean, invstd, part, P,
                       ctot, Z);
  else
    hipLaunchKernelGGL((bng_bwd_stats_kernel<false>), dim3(cbtot, Z),
                       dim3(256), 0, stream, a, mean, invstd, part, P,
                       ctot, Z);
  hipLaunchKernelGGL(bn_bwd_finalize_kernel, dim3(ctot), dim3(64), 0, stream,
                     part, (__bf16*)dgamma, (__bf16*)dbeta, s1n, s2n, ctot,
                     Z, 1.f / (float)P);
  int cmax = 0;
  for (int i = 0; i < n; ++i) cmax = Cs[i] > cmax ? Cs[i] : cmax;
  dim3 ag(bng_ew_grid(P, cmax), n), ab(256);
  if (relu)
    hipLaunchKernelGGL((bng_bwd_apply_kernel<true>), ag, ab, 0, stream, a,
                       mean, invstd, s1n, s2n, P);
  else
    hipLaunchKernelGGL((bng_bwd_apply_kernel<false>), ag, ab, 0, stream, a,
                       mean, invstd, s1n, s2n, P);
}
