// bf16 MFMA GEMM with fused epilogues (worker compute path).
//
// Replaces the reference workload's tf.matmul/xw_plus_b/relu placed on
// workers (examples/mnist/mnist_replica.py:140-143, matrix_factorization
// .py:30). CDNA4-native: v_mfma_f32_16x16x32_bf16 per-wave tiles, fp32
// accumulate, LDS-staged operand tiles, wave64 fragment layouts
// (A: lane l -> row l&15, k (l>>4)*8+j ; C/D: col l&15, row (l>>4)*4+r).
// All four op(A)/op(B) transpose combos are native (backward GEMMs
// dW = X^T dY and dX = dY W^T run without materialized transposes).
//
// Performance features (driven by profiles/r01_mnist_n1_kernel_stats.txt,
// where the un-split fwd GEMM ran 4 workgroups on a 256-CU chip):
//  * split-K: grid.z slices each store their fp32 partial tile into a
//    per-slice workspace stripe; the LAST workgroup to arrive at each
//    output tile (agent-scope arrival counter) sums the stripes in fixed
//    slice order and runs the fused bias/activation epilogue — one
//    kernel, no separate zero/reduce launches, and DETERMINISTIC
//    (fixed-order fp32 sums, no atomics on the data path).
//  * vectorized LDS staging: b128 loads when the operand's leading dim
//    and base allow, b32 otherwise; both [X,K] and [K,X] storage orders
//    have contiguous global access patterns.
//  * fused epilogues: bias add, ReLU, ReLU-backward masking by a saved
//    activation (dX GEMM), and column-sum of op(B) (bias gradients ride
//    the dW GEMM; blockIdx.y==0 workgroups see every K tile of their
//    column strip in LDS anyway).
//
// Geometry: 64x64 block tile, 4 waves (2x2), 32x32 per wave, BK=32.
#include "common.h"

namespace {

constexpr int BM = 64, BN = 64, BK = 32;
constexpr int APAD = 8;  // +16B: keeps b128 fragment reads aligned

typedef __attribute__((ext_vector_type(2))) __bf16 bf16x2;

// Stage a [X,K]-stored operand tile (row x contiguous in k) into S[x][k].
// Used for A when !TA and B when TB. 256 threads x 8 elems = 64x32 tile.
DEVINL void stage_xk(const __bf16* __restrict__ P, __bf16 (*S)[BK + APAD],
                     int x0, int k0, int X, int K, int ld, int t, int vec) {
  const int x = t >> 2;
  const int kk0 = (t & 3) * 8;
  const int gx = x0 + x;
  const int gk = k0 + kk0;
  const __bf16* src = P + (long)gx * ld + gk;
  if (gx < X && gk + 8 <= K) {
    if (vec == 8) {
      *(bf16x8*)&S[x][kk0] = *(const bf16x8*)src;
    } else if (vec == 2) {
#pragma unroll
      for (int j = 0; j < 8; j += 2)
        *(bf16x2*)&S[x][kk0 + j] = *(const bf16x2*)(src + j);
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) S[x][kk0 + j] = src[j];
    }
  } else {
#pragma unroll
    for (int j = 0; j < 8; ++j)
      S[x][kk0 + j] = (gx < X && gk + j < K) ? src[j] : (__bf16)0.f;
  }
}

// Stage a [K,X]-stored operand tile (row k contiguous in x) into S[x][k].
// Used for A when TA and B when !TB. Global access stays contiguous
// (8 consecutive x per thread); the transpose happens on the LDS write.
DEVINL void stage_kx(const __bf16* __restrict__ P, __bf16 (*S)[BK + APAD],
                     int x0, int k0, int X, int K, int ld, int t, int vec) {
  const int k = t >> 3;          // 0..31 == BK
  const int xx0 = (t & 7) * 8;   // 0..56
  const int gk = k0 + k;
  const int gx = x0 + xx0;
  const __bf16* src = P + (long)gk * ld + gx;
  if (gk < K && gx + 8 <= X) {
    if (vec == 8) {
      bf16x8 v = *(const bf16x8*)src;
#pragma unroll
      for (int j = 0; j < 8; ++j) S[xx0 + j][k] = v[j];
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) S[xx0 + j][k] = src[j];
    }
  } else {
#pragma unroll
    for (int j = 0; j < 8; ++j)
      S[xx0 + j][k] = (gk < K && gx + j < X) ? src[j] : (__bf16)0.f;
  }
}

// ACT: 0 none, 1 relu, 2 relu-bwd (mask by aux>0, the saved activation).
// SK: split-K over gridDim.z with last-arriver epilogue.
// CS: colsum_out[n] = sum_k op(B)[k][n] computed by blockIdx.y==0 WGs.
template <bool TA, bool TB, int ACT, bool BIAS, bool OUTF32, bool SK, bool CS>
__global__ __launch_bounds__(256)
void gemm_kernel(const __bf16* __restrict__ A, const __bf16* __restrict__ B,
                 const void* __restrict__ bias, bool bias_bf16,
                 void* __restrict__ Cout, const __bf16* __restrict__ aux,
                 float* __restrict__ colsum_out, float* __restrict__ ws,
                 int* __restrict__ cnt, int M, int N, int K, int lda, int ldb,
                 int ldc, int kc, int veca, int vecb) {
  __shared__ __align__(16) __bf16 As[BM][BK + APAD];   // [m][k]
  __shared__ __align__(16) __bf16 Bs[BN][BK + APAD];   // [n][k] (B^T tile)

  const int tm0 = blockIdx.y * BM;
  const int tn0 = blockIdx.x * BN;
  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;          // 0..3
  const int wr = wave >> 1, wc = wave & 1;

  const int ks = SK ? blockIdx.z * kc : 0;
  const int ke = SK ? min(ks + kc, K) : K;

  f32x4 acc[2][2] = {};
  float cs_acc = 0.f;

  for (int k0 = ks; k0 < ke; k0 += BK) {
    if (TA) stage_kx(A, As, tm0, k0, M, K, lda, t, veca);
    else    stage_xk(A, As, tm0, k0, M, K, lda, t, veca);
    if (TB) stage_xk(B, Bs, tn0, k0, N, K, ldb, t, vecb);
    else    stage_kx(B, Bs, tn0, k0, N, K, ldb, t, vecb);
    __syncthreads();

    if (CS && blockIdx.y == 0 && t < BN) {
#pragma unroll
      for (int kk = 0; kk < BK; ++kk) cs_acc += (float)Bs[t][kk];
    }

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

  if (CS && blockIdx.y == 0 && t < BN && tn0 + t < N)
    colsum_out[tn0 + t] = cs_acc;

  __shared__ int is_last;
  const long slice_stride = (long)M * ldc;
  if (SK) {
    // store this slice's fp32 partial tile into its workspace stripe
    // (plain per-slice stores -> fixed-order summation is deterministic)
    float* wslice = ws + (long)blockIdx.z * slice_stride;
#pragma unroll
    for (int fm = 0; fm < 2; ++fm)
#pragma unroll
      for (int fn = 0; fn < 2; ++fn) {
        const int col = tn0 + wc * 32 + fn * 16 + (lane & 15);
        if (col >= N) continue;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int row = tm0 + wr * 32 + fm * 16 + (lane >> 4) * 4 + r;
          if (row < M)
            __hip_atomic_store(&wslice[(long)row * ldc + col],
                               acc[fm][fn][r], __ATOMIC_RELAXED,
                               __HIP_MEMORY_SCOPE_AGENT);
        }
      }
    __threadfence();
    if (t == 0) {
      const int tile = blockIdx.y * gridDim.x + blockIdx.x;
      int old = __hip_atomic_fetch_add(&cnt[tile], 1, __ATOMIC_ACQ_REL,
                                       __HIP_MEMORY_SCOPE_AGENT);
      is_last = (old == (int)gridDim.z - 1);
    }
    __syncthreads();
    if (!is_last) return;
  }

  // epilogue: (last WG per tile when SK) bias + activation + store
  // (C/D map: col=l&15, row=(l>>4)*4+r)
#pragma unroll
  for (int fm = 0; fm < 2; ++fm) {
#pragma unroll
    for (int fn = 0; fn < 2; ++fn) {
      const int col = tn0 + wc * 32 + fn * 16 + (lane & 15);
      if (col >= N) continue;
      const float bv = !BIAS ? 0.f
          : (bias_bf16 ? (float)((const __bf16*)bias)[col]
                       : ((const float*)bias)[col]);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = tm0 + wr * 32 + fm * 16 + (lane >> 4) * 4 + r;
        if (row >= M) continue;
        const long idx = (long)row * ldc + col;
        float v;
        if (SK) {
          v = 0.f;
          for (int z = 0; z < (int)gridDim.z; ++z)
            v += __hip_atomic_load(&ws[z * slice_stride + idx],
                                   __ATOMIC_RELAXED,
                                   __HIP_MEMORY_SCOPE_AGENT);
        } else {
          v = acc[fm][fn][r];
        }
        v += bv;
        if (ACT == 1) v = v > 0.f ? v : 0.f;
        if (ACT == 2) v = (float)aux[idx] > 0.f ? v : 0.f;
        if (OUTF32)
          ((float*)Cout)[idx] = v;
        else
          ((__bf16*)Cout)[idx] = (__bf16)v;
      }
    }
  }
  if (SK && t == 0)
    __hip_atomic_store(&cnt[blockIdx.y * gridDim.x + blockIdx.x], 0,
                       __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
}

}  // namespace

void launch_gemm(const bf16_t* A, const bf16_t* B, const void* bias,
                 bool bias_bf16, void* C, bool out_f32, const bf16_t* aux,
                 float* colsum_out, float* ws, int* cnt, int kc, int nslice,
                 int M, int N, int K, int lda, int ldb, int ldc, bool ta,
                 bool tb, int act, int veca, int vecb, hipStream_t stream) {
  const bool sk = nslice > 1;
  dim3 grid(ceil_div(N, BN), ceil_div(M, BM), sk ? nslice : 1);
  dim3 block(256);
  const bool has_bias = bias != nullptr;
  const bool cs = colsum_out != nullptr;
#define LAUNCH(TAv, TBv, ACTv, BIASv, OUTv, SKv, CSv)                       \
  hipLaunchKernelGGL((gemm_kernel<TAv, TBv, ACTv, BIASv, OUTv, SKv, CSv>),  \
                     grid, block, 0, stream, (const __bf16*)A,              \
                     (const __bf16*)B, bias, bias_bf16, C,                  \
                     (const __bf16*)aux, colsum_out, ws, cnt, M, N, K, lda, \
                     ldb, ldc, kc, veca, vecb)
  // constraint combos (relu_bwd only nt/bf16, colsum only tn, split-K only
  // nn) are validated by the bindings in ext.hip
  if (act == 2) {
    LAUNCH(false, true, 2, false, false, false, false);
    return;
  }
  if (cs) {
    if (out_f32) LAUNCH(true, false, 0, false, true, false, true);
    else         LAUNCH(true, false, 0, false, false, false, true);
    return;
  }
  if (sk) {
#define SK_ACT(ACTv)                                                        \
    do {                                                                    \
      if (has_bias) { if (out_f32) LAUNCH(false, false, ACTv, true, true,   \
                                          true, false);                     \
                      else LAUNCH(false, false, ACTv, true, false, true,    \
                                  false); }                                 \
      else          { if (out_f32) LAUNCH(false, false, ACTv, false, true,  \
                                          true, false);                     \
                      else LAUNCH(false, false, ACTv, false, false, true,   \
                                  false); }                                 \
    } while (0)
    if (act == 1) SK_ACT(1); else SK_ACT(0);
#undef SK_ACT
    return;
  }
#define DISP_OUT(TAv, TBv, ACTv, BIASv)                                     \
  do { if (out_f32) LAUNCH(TAv, TBv, ACTv, BIASv, true, false, false);      \
       else LAUNCH(TAv, TBv, ACTv, BIASv, false, false, false); } while (0)
#define DISP_BIAS(TAv, TBv, ACTv)                                           \
  do { if (has_bias) DISP_OUT(TAv, TBv, ACTv, true);                        \
       else DISP_OUT(TAv, TBv, ACTv, false); } while (0)
#define DISP_ACT(TAv, TBv)                                                  \
  do { if (act == 1) DISP_BIAS(TAv, TBv, 1); else DISP_BIAS(TAv, TBv, 0); } while (0)
  if (ta) { if (tb) DISP_ACT(true, true); else DISP_ACT(true, false); }
  else    { if (tb) DISP_ACT(false, true); else DISP_ACT(false, false); }
#undef DISP_ACT
#undef DISP_BIAS
#undef DISP_OUT
#undef LAUNCH
}
