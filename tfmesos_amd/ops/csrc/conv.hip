// Implicit-GEMM 2-D convolution on MFMA (bf16 in, fp32 accumulate).
//
// The Inception-v3 config of BASELINE.json needs conv fwd/bwd on the
// worker compute path (the reference delegated convs to TF kernels;
// SURVEY.md §2b row "conv stack"). CDNA4-native design: the conv is a
// GEMM over [output pixels] x [filter taps], with the im2col patch
// gather fused into the LDS staging (no materialized im2col buffer —
// Inception's mid layers would need >100 GB of col data), and the same
// v_mfma_f32_16x16x32_bf16 64x64 core as gemm.hip. NCHW layout.
//
//   fwd:    Y[n,k,ho,wo]  = sum_{c,r,s} X[n,c,ho*U+r-P, wo*V+s-Q] W[k,c,r,s]
//           GEMM  M = N*Ho*Wo, Ncol = K,     Kdim = C*R*S
//   bwd-d:  dX[n,c,h,w]   = sum_{k,r,s} dY[n,k,(h+P-r)/U,(w+Q-s)/V] W[k,c,r,s]
//           GEMM  M = N*H*W,   Ncol = C,     Kdim = K*R*S   (U,V-divisible taps)
//   bwd-w:  dW[k,c,r,s]   = sum_{n,ho,wo} dY[n,k,ho,wo] X[n,c,ho*U+r-P,...]
//           GEMM  M = K,       Ncol = C*R*S, Kdim = N*Ho*Wo
//
// Staging performance: every gather decodes its pixel index ONCE per
// 8-element strip (integer division is the expensive op — the first,
// correctness-only version did 8 long divisions per strip and ran at
// <1% MFMA peak, profiles/r01_inception_n1_kernel_stats.txt), then
// walks contiguous addresses; full-row interior strips take a straight
// pointer walk, boundary strips an incremental-carry path.
#include "common.h"

namespace {

constexpr int BM = 64, BN = 64, BK = 32;
constexpr int APAD = 8;

struct ConvShape {
  int N, C, H, W;     // input
  int K, R, S;        // filter
  int Ho, Wo;         // output
  int U, V;           // stride
  int P, Q;           // pad
};

// ---------------------------------------------------------------- forward

// Stage a 64(m) x 32(k) patch tile: m = output pixel, k = (c,r,s) tap.
// Thread t loads 8 consecutive m (-> consecutive wo) for one tap.
DEVINL void stage_patch_fwd(const __bf16* __restrict__ X, __bf16 (*Sm)[BK + APAD],
                            const ConvShape cs, long m0, int k0, long M,
                            int CRS, int t) {
  const int kk = t >> 3;          // 0..31
  const int mm0 = (t & 7) * 8;
  const int q = k0 + kk;
  const long pm0 = m0 + mm0;
  if (q >= CRS || pm0 >= M) {
#pragma unroll
    for (int j = 0; j < 8; ++j) Sm[mm0 + j][kk] = (__bf16)0.f;
    return;
  }
  const int c = q / (cs.R * cs.S);
  const int rs = q - c * (cs.R * cs.S);
  const int r = rs / cs.S;
  const int s = rs - r * cs.S;
  // decode pixel once (32-bit divisions; M < 2^31 enforced by host)
  const int HoWo = cs.Ho * cs.Wo;
  int n = (int)(pm0 / HoWo);
  int rem = (int)(pm0 - (long)n * HoWo);
  int ho = rem / cs.Wo;
  int wo = rem - ho * cs.Wo;
  const int hi = ho * cs.U + r - cs.P;
  const int wi0 = wo * cs.V + s - cs.Q;

  if (wo + 8 <= cs.Wo && pm0 + 8 <= M &&
      hi >= 0 && hi < cs.H && wi0 >= 0 && wi0 + 7 * cs.V < cs.W) {
    // fast path: one row, fully interior — straight pointer walk
    const __bf16* src = X + (((long)n * cs.C + c) * cs.H + hi) * cs.W + wi0;
    if (cs.V == 1) {
#pragma unroll
      for (int j = 0; j < 8; ++j) Sm[mm0 + j][kk] = src[j];
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) Sm[mm0 + j][kk] = src[j * cs.V];
    }
    return;
  }
  // slow path: incremental carry across wo/ho/n, per-element bounds
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    float v = 0.f;
    if (pm0 + j < M) {
      const int hij = ho * cs.U + r - cs.P;
      const int wij = wo * cs.V + s - cs.Q;
      if (hij >= 0 && hij < cs.H && wij >= 0 && wij < cs.W)
        v = (float)X[(((long)n * cs.C + c) * cs.H + hij) * cs.W + wij];
    }
    Sm[mm0 + j][kk] = (__bf16)v;
    if (++wo == cs.Wo) { wo = 0; if (++ho == cs.Ho) { ho = 0; ++n; } }
  }
}

// Stage the weight tile Bs[kchan][tap]: W flat [K][CRS], row-contiguous.
DEVINL void stage_wtile(const __bf16* __restrict__ Wt, __bf16 (*Sn)[BK + APAD],
                        int n0, int k0, int K, int CRS, int t) {
  const int x = t >> 2;
  const int kk0 = (t & 3) * 8;
  const int gx = n0 + x;
  const int gk = k0 + kk0;
  const __bf16* src = Wt + (long)gx * CRS + gk;
  if (gx < K && gk + 8 <= CRS) {
#pragma unroll
    for (int j = 0; j < 8; ++j) Sn[x][kk0 + j] = src[j];
  } else {
#pragma unroll
    for (int j = 0; j < 8; ++j)
      Sn[x][kk0 + j] = (gx < K && gk + j < CRS) ? src[j] : (__bf16)0.f;
  }
}

template <bool BIAS, bool RELU>
__global__ __launch_bounds__(256)
void conv_fwd_kernel(const __bf16* __restrict__ X, const __bf16* __restrict__ Wt,
                     const float* __restrict__ bias, __bf16* __restrict__ Y,
                     ConvShape cs) {
  __shared__ __align__(16) __bf16 As[BM][BK + APAD];
  __shared__ __align__(16) __bf16 Bs[BN][BK + APAD];
  const long M = (long)cs.N * cs.Ho * cs.Wo;
  const int CRS = cs.C * cs.R * cs.S;
  const long tm0 = (long)blockIdx.x * BM;   // pixel tiles ride grid.x (2^31)
  const int tn0 = blockIdx.y * BN;
  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const int wr = wave >> 1, wc = wave & 1;

  f32x4 acc[2][2] = {};
  for (int k0 = 0; k0 < CRS; k0 += BK) {
    stage_patch_fwd(X, As, cs, tm0, k0, M, CRS, t);
    stage_wtile(Wt, Bs, tn0, k0, cs.K, CRS, t);
    __syncthreads();
    const int kfrag = (lane >> 4) * 8;
#pragma unroll
    for (int fm = 0; fm < 2; ++fm) {
      bf16x8 a = *(const bf16x8*)&As[wr * 32 + fm * 16 + (lane & 15)][kfrag];
#pragma unroll
      for (int fn = 0; fn < 2; ++fn) {
        bf16x8 b = *(const bf16x8*)&Bs[wc * 32 + fn * 16 + (lane & 15)][kfrag];
        acc[fm][fn] =
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[fm][fn], 0, 0, 0);
      }
    }
    __syncthreads();
  }

  // epilogue: Y[n,k,ho,wo] (col = k channel, row = output pixel)
#pragma unroll
  for (int fm = 0; fm < 2; ++fm)
#pragma unroll
    for (int fn = 0; fn < 2; ++fn) {
      const int k = tn0 + wc * 32 + fn * 16 + (lane & 15);
      if (k >= cs.K) continue;
      const float bv = BIAS ? bias[k] : 0.f;
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        const long pm = tm0 + wr * 32 + fm * 16 + (lane >> 4) * 4 + rr;
        if (pm >= M) continue;
        const int n = (int)(pm / (cs.Ho * cs.Wo));
        const int rem = (int)(pm % (cs.Ho * cs.Wo));
        float v = acc[fm][fn][rr] + bv;
        if (RELU) v = v > 0.f ? v : 0.f;
        Y[(((long)n * cs.K + k) * cs.Ho + rem / cs.Wo) * cs.Wo + rem % cs.Wo] =
            (__bf16)v;
      }
    }
}

// --------------------------------------------------------------- bwd-data

// m = input pixel (n,h,w); tap q = (k,r,s); contributes when
// (h+P-r) % U == 0 and in range (same for w). STRIDE1 specializes the
// common U==V==1 case (no divisibility tests, contiguous fast path).
template <bool STRIDE1>
DEVINL void stage_patch_bwdd(const __bf16* __restrict__ dY,
                             __bf16 (*Sm)[BK + APAD], const ConvShape cs,
                             long m0, int k0, long M, int KRS, int t) {
  const int kk = t >> 3;
  const int mm0 = (t & 7) * 8;
  const int q = k0 + kk;
  const long pm0 = m0 + mm0;
  if (q >= KRS || pm0 >= M) {
#pragma unroll
    for (int j = 0; j < 8; ++j) Sm[mm0 + j][kk] = (__bf16)0.f;
    return;
  }
  const int k = q / (cs.R * cs.S);
  const int rs = q - k * (cs.R * cs.S);
  const int r = rs / cs.S;
  const int s = rs - r * cs.S;
  const int HWi = cs.H * cs.W;
  int n = (int)(pm0 / HWi);
  int rem = (int)(pm0 - (long)n * HWi);
  int h = rem / cs.W;
  int w = rem - h * cs.W;

  if (STRIDE1) {
    const int ho = h + cs.P - r;
    const int wo0 = w + cs.Q - s;
    if (w + 8 <= cs.W && pm0 + 8 <= M &&
        ho >= 0 && ho < cs.Ho && wo0 >= 0 && wo0 + 8 <= cs.Wo) {
      const __bf16* src =
          dY + (((long)n * cs.K + k) * cs.Ho + ho) * cs.Wo + wo0;
#pragma unroll
      for (int j = 0; j < 8; ++j) Sm[mm0 + j][kk] = src[j];
      return;
    }
  }
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    float v = 0.f;
    if (pm0 + j < M) {
      const int hn = h + cs.P - r, wn = w + cs.Q - s;
      if (STRIDE1) {
        if (hn >= 0 && hn < cs.Ho && wn >= 0 && wn < cs.Wo)
          v = (float)dY[(((long)n * cs.K + k) * cs.Ho + hn) * cs.Wo + wn];
      } else {
        if (hn >= 0 && wn >= 0 && hn % cs.U == 0 && wn % cs.V == 0) {
          const int ho = hn / cs.U, wo = wn / cs.V;
          if (ho < cs.Ho && wo < cs.Wo)
            v = (float)dY[(((long)n * cs.K + k) * cs.Ho + ho) * cs.Wo + wo];
        }
      }
    }
    Sm[mm0 + j][kk] = (__bf16)v;
    if (++w == cs.W) { w = 0; if (++h == cs.H) { h = 0; ++n; } }
  }
}

// Weight tile for bwd-data: Bs[c][tap(k,r,s)] = W[k,c,r,s] (strided gather;
// taps decode incrementally — consecutive q walk s, then r, then k).
DEVINL void stage_wtile_bwdd(const __bf16* __restrict__ Wt,
                             __bf16 (*Sn)[BK + APAD], const ConvShape cs,
                             int n0, int k0, int KRS, int t) {
  const int x = t >> 2;          // c offset 0..63
  const int kk0 = (t & 3) * 8;
  const int c = n0 + x;
  const int q0 = k0 + kk0;
  if (c >= cs.C || q0 >= KRS) {
#pragma unroll
    for (int j = 0; j < 8; ++j) Sn[x][kk0 + j] = (__bf16)0.f;
    return;
  }
  const int RS = cs.R * cs.S;
  int k = q0 / RS;
  int rs = q0 - k * RS;
  int r = rs / cs.S;
  int s = rs - r * cs.S;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    float v = 0.f;
    if (q0 + j < KRS)
      v = (float)Wt[(((long)k * cs.C + c) * cs.R + r) * cs.S + s];
    Sn[x][kk0 + j] = (__bf16)v;
    if (++s == cs.S) { s = 0; if (++r == cs.R) { r = 0; ++k; } }
  }
}

template <bool STRIDE1>
__global__ __launch_bounds__(256)
void conv_bwdd_kernel(const __bf16* __restrict__ dY, const __bf16* __restrict__ Wt,
                      __bf16* __restrict__ dX, ConvShape cs) {
  __shared__ __align__(16) __bf16 As[BM][BK + APAD];
  __shared__ __align__(16) __bf16 Bs[BN][BK + APAD];
  const long M = (long)cs.N * cs.H * cs.W;
  const int KRS = cs.K * cs.R * cs.S;
  const long tm0 = (long)blockIdx.x * BM;   // pixel tiles ride grid.x (2^31)
  const int tn0 = blockIdx.y * BN;
  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const int wr = wave >> 1, wc = wave & 1;

  f32x4 acc[2][2] = {};
  for (int k0 = 0; k0 < KRS; k0 += BK) {
    stage_patch_bwdd<STRIDE1>(dY, As, cs, tm0, k0, M, KRS, t);
    stage_wtile_bwdd(Wt, Bs, cs, tn0, k0, KRS, t);
    __syncthreads();
    const int kfrag = (lane >> 4) * 8;
#pragma unroll
    for (int fm = 0; fm < 2; ++fm) {
      bf16x8 a = *(const bf16x8*)&As[wr * 32 + fm * 16 + (lane & 15)][kfrag];
#pragma unroll
      for (int fn = 0; fn < 2; ++fn) {
        bf16x8 b = *(const bf16x8*)&Bs[wc * 32 + fn * 16 + (lane & 15)][kfrag];
        acc[fm][fn] =
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[fm][fn], 0, 0, 0);
      }
    }
    __syncthreads();
  }

#pragma unroll
  for (int fm = 0; fm < 2; ++fm)
#pragma unroll
    for (int fn = 0; fn < 2; ++fn) {
      const int c = tn0 + wc * 32 + fn * 16 + (lane & 15);
      if (c >= cs.C) continue;
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        const long pm = tm0 + wr * 32 + fm * 16 + (lane >> 4) * 4 + rr;
        if (pm >= M) continue;
        const int n = (int)(pm / ((long)cs.H * cs.W));
        const int rem = (int)(pm % ((long)cs.H * cs.W));
        dX[(((long)n * cs.C + c) * cs.H + rem / cs.W) * cs.W + rem % cs.W] =
            (__bf16)acc[fm][fn][rr];
      }
    }
}

// -------------------------------------------------------------- bwd-weight

// GEMM: rows m = output channel k (M=K), cols = (c,r,s), Kdim = N*Ho*Wo.
// As[k][p] = dY[n,k,ho,wo]; thread t loads 8 consecutive p for one k
// (contiguous in dY within an image).
DEVINL void stage_dy_bwdw(const __bf16* __restrict__ dY, __bf16 (*Sm)[BK + APAD],
                          const ConvShape cs, int m0, long p0, long Ptot,
                          int t) {
  const int pp0 = (t & 3) * 8;     // p offset 0..24
  const int kx = t >> 2;           // k-channel row 0..63
  const int k = m0 + kx;
  const long p = p0 + pp0;
  const long HoWo = (long)cs.Ho * cs.Wo;
  if (k >= cs.K || p >= Ptot) {
#pragma unroll
    for (int j = 0; j < 8; ++j) Sm[kx][pp0 + j] = (__bf16)0.f;
    return;
  }
  int n = (int)(p / HoWo);
  int rem = (int)(p - (long)n * HoWo);
  if (p + 8 <= Ptot && rem + 8 <= (int)HoWo) {
    const __bf16* src = dY + ((long)n * cs.K + k) * HoWo + rem;
#pragma unroll
    for (int j = 0; j < 8; ++j) Sm[kx][pp0 + j] = src[j];
    return;
  }
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    float v = 0.f;
    if (p + j < Ptot)
      v = (float)dY[((long)n * cs.K + k) * HoWo + rem];
    Sm[kx][pp0 + j] = (__bf16)v;
    if (++rem == (int)HoWo) { rem = 0; ++n; }
  }
}

// Bs[crs][p] = X[n, c, ho*U+r-P, wo*V+s-Q]; 8 consecutive p for one tap.
DEVINL void stage_x_bwdw(const __bf16* __restrict__ X, __bf16 (*Sn)[BK + APAD],
                         const ConvShape cs, int n0, long p0, int CRS,
                         long Ptot, int t) {
  const int pp0 = (t & 3) * 8;
  const int qx = t >> 2;           // tap row 0..63
  const int q = n0 + qx;
  const long p = p0 + pp0;
  if (q >= CRS || p >= Ptot) {
#pragma unroll
    for (int j = 0; j < 8; ++j) Sn[qx][pp0 + j] = (__bf16)0.f;
    return;
  }
  const int c = q / (cs.R * cs.S);
  const int rs = q - c * (cs.R * cs.S);
  const int r = rs / cs.S;
  const int s = rs - r * cs.S;
  const long HoWo = (long)cs.Ho * cs.Wo;
  int n = (int)(p / HoWo);
  int rem = (int)(p - (long)n * HoWo);
  int ho = rem / cs.Wo;
  int wo = rem - ho * cs.Wo;
  const int hi = ho * cs.U + r - cs.P;
  const int wi0 = wo * cs.V + s - cs.Q;
  if (p + 8 <= Ptot && wo + 8 <= cs.Wo &&
      hi >= 0 && hi < cs.H && wi0 >= 0 && wi0 + 7 * cs.V < cs.W) {
    const __bf16* src = X + (((long)n * cs.C + c) * cs.H + hi) * cs.W + wi0;
    if (cs.V == 1) {
#pragma unroll
      for (int j = 0; j < 8; ++j) Sn[qx][pp0 + j] = src[j];
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) Sn[qx][pp0 + j] = src[j * cs.V];
    }
    return;
  }
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    float v = 0.f;
    if (p + j < Ptot) {
      const int hij = ho * cs.U + r - cs.P;
      const int wij = wo * cs.V + s - cs.Q;
      if (hij >= 0 && hij < cs.H && wij >= 0 && wij < cs.W)
        v = (float)X[(((long)n * cs.C + c) * cs.H + hij) * cs.W + wij];
    }
    Sn[qx][pp0 + j] = (__bf16)v;
    if (++wo == cs.Wo) { wo = 0; if (++ho == cs.Ho) { ho = 0; ++n; } }
  }
}

// dW fp32 out [K][C*R*S]; grid.z slices the huge N*Ho*Wo reduction and
// accumulates with fp32 atomics (dW is zeroed by the launcher).
__global__ __launch_bounds__(256)
void conv_bwdw_kernel(const __bf16* __restrict__ dY, const __bf16* __restrict__ X,
                      float* __restrict__ dW, ConvShape cs, long pc) {
  __shared__ __align__(16) __bf16 As[BM][BK + APAD];
  __shared__ __align__(16) __bf16 Bs[BN][BK + APAD];
  const int CRS = cs.C * cs.R * cs.S;
  const long Ptot = (long)cs.N * cs.Ho * cs.Wo;
  const int tm0 = blockIdx.y * BM;
  const int tn0 = blockIdx.x * BN;
  const long ps = (long)blockIdx.z * pc;
  const long pe = min(ps + pc, Ptot);
  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const int wr = wave >> 1, wc = wave & 1;

  f32x4 acc[2][2] = {};
  for (long p0 = ps; p0 < pe; p0 += BK) {
    stage_dy_bwdw(dY, As, cs, tm0, p0, Ptot, t);
    stage_x_bwdw(X, Bs, cs, tn0, p0, CRS, Ptot, t);
    __syncthreads();
    const int kfrag = (lane >> 4) * 8;
#pragma unroll
    for (int fm = 0; fm < 2; ++fm) {
      bf16x8 a = *(const bf16x8*)&As[wr * 32 + fm * 16 + (lane & 15)][kfrag];
#pragma unroll
      for (int fn = 0; fn < 2; ++fn) {
        bf16x8 b = *(const bf16x8*)&Bs[wc * 32 + fn * 16 + (lane & 15)][kfrag];
        acc[fm][fn] =
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[fm][fn], 0, 0, 0);
      }
    }
    __syncthreads();
  }

#pragma unroll
  for (int fm = 0; fm < 2; ++fm)
#pragma unroll
    for (int fn = 0; fn < 2; ++fn) {
      const int q = tn0 + wc * 32 + fn * 16 + (lane & 15);
      if (q >= CRS) continue;
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        const int k = tm0 + wr * 32 + fm * 16 + (lane >> 4) * 4 + rr;
        if (k >= cs.K) continue;
        if (gridDim.z == 1)
          dW[(long)k * CRS + q] = acc[fm][fn][rr];
        else
          unsafeAtomicAdd(&dW[(long)k * CRS + q], acc[fm][fn][rr]);
      }
    }
}

}  // namespace

void launch_conv_fwd(const bf16_t* X, const bf16_t* W, const float* bias,
                     bf16_t* Y, int N, int C, int H, int Wd, int K, int R,
                     int S, int Ho, int Wo, int U, int V, int P, int Q,
                     bool relu, hipStream_t stream) {
  ConvShape cs{N, C, H, Wd, K, R, S, Ho, Wo, U, V, P, Q};
  const long M = (long)N * Ho * Wo;
  dim3 grid((unsigned)((M + BM - 1) / BM), ceil_div(K, BN));
  dim3 block(256);
  if (bias) {
    if (relu)
      hipLaunchKernelGGL((conv_fwd_kernel<true, true>), grid, block, 0,
                         stream, (const __bf16*)X, (const __bf16*)W, bias,
                         (__bf16*)Y, cs);
    else
      hipLaunchKernelGGL((conv_fwd_kernel<true, false>), grid, block, 0,
                         stream, (const __bf16*)X, (const __bf16*)W, bias,
                         (__bf16*)Y, cs);
  } else {
    if (relu)
      hipLaunchKernelGGL((conv_fwd_kernel<false, true>), grid, block, 0,
                         stream, (const __bf16*)X, (const __bf16*)W, bias,
                         (__bf16*)Y, cs);
    else
      hipLaunchKernelGGL((conv_fwd_kernel<false, false>), grid, block, 0,
                         stream, (const __bf16*)X, (const __bf16*)W, bias,
                         (__bf16*)Y, cs);
  }
}

void launch_conv_bwd_data(const bf16_t* dY, const bf16_t* W, bf16_t* dX,
                          int N, int C, int H, int Wd, int K, int R, int S,
                          int Ho, int Wo, int U, int V, int P, int Q,
                          hipStream_t stream) {
  ConvShape cs{N, C, H, Wd, K, R, S, Ho, Wo, U, V, P, Q};
  const long M = (long)N * H * Wd;
  dim3 grid((unsigned)((M + BM - 1) / BM), ceil_div(C, BN));
  dim3 block(256);
  if (U == 1 && V == 1)
    hipLaunchKernelGGL((conv_bwdd_kernel<true>), grid, block, 0, stream,
                       (const __bf16*)dY, (const __bf16*)W, (__bf16*)dX, cs);
  else
    hipLaunchKernelGGL((conv_bwdd_kernel<false>), grid, block, 0, stream,
                       (const __bf16*)dY, (const __bf16*)W, (__bf16*)dX, cs);
}

void launch_conv_bwd_weight(const bf16_t* dY, const bf16_t* X, float* dW,
                            int N, int C, int H, int Wd, int K, int R, int S,
                            int Ho, int Wo, int U, int V, int P, int Q,
                            hipStream_t stream) {
  ConvShape cs{N, C, H, Wd, K, R, S, Ho, Wo, U, V, P, Q};
  const int CRS = C * R * S;
  const long Ptot = (long)N * Ho * Wo;
  // slice the reduction so the grid can fill the chip (>=512 WGs)
  const long tiles = (long)ceil_div(K, BM) * ceil_div(CRS, BN);
  long zmax = (Ptot + BK - 1) / BK;
  long zwant = 512 / tiles;
  if (zwant < 1) zwant = 1;
  int z = (int)(zmax < zwant ? zmax : zwant);
  long pc = (Ptot + z - 1) / z;
  pc = (pc + BK - 1) / BK * BK;
  z = (int)((Ptot + pc - 1) / pc);
  dim3 grid(ceil_div(CRS, BN), ceil_div(K, BM), z);
  dim3 block(256);
  hipLaunchKernelGGL(conv_bwdw_kernel, grid, block, 0, stream,
                     (const __bf16*)dY, (const __bf16*)X, dW, cs, pc);
}
