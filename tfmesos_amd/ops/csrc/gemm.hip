// bf16 MFMA GEMM with fused bias+ReLU epilogue (worker compute path).
//
// Replaces the reference workload's tf.matmul/xw_plus_b/relu placed on
// workers (examples/mnist/mnist_replica.py:140-143, matrix_factorization
// .py:30). CDNA4-native: v_mfma_f32_16x16x32_bf16 per-wave tiles, fp32
// accumulate, LDS-staged operand tiles, wave64 fragment layouts
// (A: lane l -> row l&15, k (l>>4)*8+j ; C/D: col l&15, row (l>>4)*4+r).
// All four op(A)/op(B) transpose combos are native (backward GEMMs
// dW = X^T dY and dX = dY W^T run without materialized transposes).
// Output dtype bf16 (activations) or fp32 (gradients written straight
// into the PS flat grad buffer).
//
// Geometry: 64x64 block tile, 4 waves (2x2), 32x32 per wave, BK=32.
// Sized for the reference's small/mid shapes; correctness-first with
// vectorized LDS fragment reads; tune pass comes after rocprof.
#include "common.h"

namespace {

constexpr int BM = 64, BN = 64, BK = 32;
constexpr int APAD = 8;  // +16B: keeps b128 fragment reads aligned

template <bool TA, bool TB, int ACT, bool BIAS, bool OUTF32>
__global__ __launch_bounds__(256)
void gemm_kernel(const bf16_t* __restrict__ A, const bf16_t* __restrict__ B,
                 const void* __restrict__ bias, bool bias_bf16,
                 void* __restrict__ Cout,
                 int M, int N, int K, int lda, int ldb, int ldc) {
  __shared__ bf16_t As[BM][BK + APAD];   // [m][k]
  __shared__ bf16_t Bs[BN][BK + APAD];   // [n][k] (B^T tile)

  const int tm0 = blockIdx.y * BM;
  const int tn0 = blockIdx.x * BN;
  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;          // 0..3
  const int wr = wave >> 1, wc = wave & 1;

  f32x4 acc[2][2] = {};

  for (int k0 = 0; k0 < K; k0 += BK) {
    // stage A tile: thread t loads 8 elements of row (t>>2)
    {
      const int m = t >> 2;
      const int kk0 = (t & 3) * 8;
      const int gm = tm0 + m;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int gk = k0 + kk0 + j;
        float v = 0.f;
        if (gm < M && gk < K)
          v = bf2f(TA ? A[(long)gk * lda + gm] : A[(long)gm * lda + gk]);
        As[m][kk0 + j] = f2bf(v);
      }
    }
    // stage B^T tile: thread t loads 8 elements of col (t>>2)
    {
      const int n = t >> 2;
      const int kk0 = (t & 3) * 8;
      const int gn = tn0 + n;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int gk = k0 + kk0 + j;
        float v = 0.f;
        if (gn < N && gk < K)
          v = bf2f(TB ? B[(long)gn * ldb + gk] : B[(long)gk * ldb + gn]);
        Bs[n][kk0 + j] = f2bf(v);
      }
    }
    __syncthreads();

    // two K-steps of 16x16x32 MFMA per tile? No: BK==32 == the MFMA K.
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

  // epilogue: bias + activation + store (C/D map: col=l&15, row=(l>>4)*4+r)
#pragma unroll
  for (int fm = 0; fm < 2; ++fm) {
#pragma unroll
    for (int fn = 0; fn < 2; ++fn) {
      const int col = tn0 + wc * 32 + fn * 16 + (lane & 15);
      if (col >= N) continue;
      const float bv = !BIAS ? 0.f
          : (bias_bf16 ? bf2f(((const bf16_t*)bias)[col])
                       : ((const float*)bias)[col]);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = tm0 + wr * 32 + fm * 16 + (lane >> 4) * 4 + r;
        if (row >= M) continue;
        float v = acc[fm][fn][r] + bv;
        if (ACT == 1) v = v > 0.f ? v : 0.f;
        if (OUTF32)
          ((float*)Cout)[(long)row * ldc + col] = v;
        else
          ((bf16_t*)Cout)[(long)row * ldc + col] = f2bf(v);
      }
    }
  }
}

}  // namespace

void launch_gemm(const bf16_t* A, const bf16_t* B, const void* bias,
                 bool bias_bf16, void* C, bool out_f32, int M, int N, int K,
                 int lda, int ldb, int ldc, bool ta, bool tb, int act,
                 hipStream_t stream) {
  dim3 grid(ceil_div(N, BN), ceil_div(M, BM));
  dim3 block(256);
  const bool has_bias = bias != nullptr;
#define DISP(TAv, TBv, ACTv, BIASv, OUTv)                                  \
  hipLaunchKernelGGL((gemm_kernel<TAv, TBv, ACTv, BIASv, OUTv>), grid,     \
                     block, 0, stream, A, B, bias, bias_bf16, C, M, N, K,   \
                     lda, ldb, ldc)
#define DISP_OUT(TAv, TBv, ACTv, BIASv)                                    \
  do { if (out_f32) DISP(TAv, TBv, ACTv, BIASv, true);                     \
       else DISP(TAv, TBv, ACTv, BIASv, false); } while (0)
#define DISP_BIAS(TAv, TBv, ACTv)                                          \
  do { if (has_bias) DISP_OUT(TAv, TBv, ACTv, true);                       \
       else DISP_OUT(TAv, TBv, ACTv, false); } while (0)
#define DISP_ACT(TAv, TBv)                                                 \
  do { if (act == 1) DISP_BIAS(TAv, TBv, 1); else DISP_BIAS(TAv, TBv, 0); } while (0)
  if (ta) { if (tb) DISP_ACT(true, true); else DISP_ACT(true, false); }
  else    { if (tb) DISP_ACT(false, true); else DISP_ACT(false, false); }
#undef DISP_ACT
#undef DISP_BIAS
#undef DISP_OUT
#undef DISP
}
