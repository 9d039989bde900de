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
//    per-slice workspace stripe (plain stores, no atomics); a small
//    vectorized reduce kernel then sums the stripes in fixed slice order
//    and applies the bias/activation epilogue. Two launches, but the
//    reduce is coalesced cached f32x4 traffic and the kernel boundary is
//    the acquire fence — DETERMINISTIC and memory-model-clean (an
//    in-kernel last-arriver tail measured 3x slower: its uncached
//    per-element loads serialize).
//  * vectorized LDS staging: b128 loads when the operand's leading dim
//    and base allow, b32 otherwise; both [X,K] and [K,X] storage orders
//    have contiguous global access patterns.
//  * fused epilogues: bias add, ReLU, ReLU-backward masking by a saved
//    activation (dX GEMM), and column-sum of op(B) (bias gradients ride
//    the dW GEMM; blockIdx.y==0 workgroups see every K tile of their
//    column strip in LDS anyway).
//
// Geometry: 64x64 block tile, 4 waves (2x2), 32x32 per wave, BK=32.
#include <cstdlib>

#include "common.h"

namespace {

constexpr int BM = 64, BN = 64, BK = 32;
// XOR-swizzled [64][32] tile for the 64x64 kernel's xk-staged
// operands (same scheme as conv.hip: stride exactly 32 elems, 8-elem
// chunk index XOR (r>>2)&3 — conflict-free for ds_read_b128's
// non-contiguous 16-lane groups AND the b128 staging writes; verified
// zero SQ_LDS_BANK_CONFLICT on the conv twins). gemm_small reuses
// the same helper for its wave-private 32x32 images.
constexpr int SWZ_ELEMS = BM * BK;

DEVINL __bf16* sptr(__bf16* S, int r, int c) {
  return S + r * BK + ((((c >> 3) ^ ((r >> 2) & 3)) << 3) | (c & 7));
}
DEVINL const __bf16* sptr(const __bf16* S, int r, int c) {
  return S + r * BK + ((((c >> 3) ^ ((r >> 2) & 3)) << 3) | (c & 7));
}

typedef __attribute__((ext_vector_type(2))) __bf16 bf16x2;

// Stage a [X,K]-stored operand tile (row x contiguous in k) into S[x][k].
// Used for A when !TA and B when TB. 256 threads x 8 elems = 64x32 tile.
DEVINL void stage_xk(const __bf16* __restrict__ P, __bf16* S,
                     int x0, int k0, int X, int K, int ld, int t, int vec) {
  const int x = t >> 2;
  const int kk0 = (t & 3) * 8;
  const int gx = x0 + x;
  const int gk = k0 + kk0;
  const __bf16* src = P + (long)gx * ld + gk;
  if (gx < X && gk + 8 <= K) {
    if (vec == 8) {
      *(bf16x8*)sptr(S, x, kk0) = *(const bf16x8*)src;
    } else if (vec == 2) {
#pragma unroll
      for (int j = 0; j < 8; j += 2)
        *(bf16x2*)sptr(S, x, kk0 + j) = *(const bf16x2*)(src + j);
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) *sptr(S, x, kk0 + j) = src[j];
    }
  } else {
#pragma unroll
    for (int j = 0; j < 8; ++j)
      *sptr(S, x, kk0 + j) = (gx < X && gk + j < K) ? src[j] : (__bf16)0.f;
  }
}

// Stage a [K,X]-stored operand tile (row k contiguous in x) into the
// reduction-major transpose-read image S[k][x ^ 16*((k>>3)&1)].
// Used for A when TA and B when !TB. The earlier k-major form paid 8
// sub-dword LDS scatter writes per thread to transpose here; this one
// is a single b128 store, and the MFMA fragments come back via
// ds_read_b64_tr_b16 (semantics measured on-device, tools/trprobe.hip
// — see conv.hip's transpose-read section). The 16-column XOR per
// k-octet makes the two k-octets a 32-lane tr read touches land on
// disjoint bank octets (16x16x32 fragments read rows kfrag..kfrag+7,
// and rows r and r+8 alias banks at the 96-elem row stride).
constexpr int TR_L = 96;
constexpr int TR_ELEMS = BK * TR_L;

DEVINL bf16x4 tr_read(const __bf16* p) {
  return __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
      (__attribute__((address_space(3))) bf16x4*)p);
}

DEVINL int tr_xor(int k) { return 16 * ((k >> 3) & 1); }

DEVINL void stage_kx(const __bf16* __restrict__ P, __bf16* S,
                     int x0, int k0, int X, int K, int ld, int t, int vec) {
  const int k = t >> 3;          // 0..31 == BK
  const int xx0 = (t & 7) * 8;   // 0..56
  const int gk = k0 + k;
  const int gx = x0 + xx0;
  const __bf16* src = P + (long)gk * ld + gx;
  bf16x8 v = {};
  if (gk < K && gx + 8 <= X) {
    if (vec == 8) {
      v = *(const bf16x8*)src;
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) v[j] = src[j];
    }
  } else if (gk < K) {
#pragma unroll
    for (int j = 0; j < 8; ++j)
      if (gx + j < X) v[j] = src[j];
  }
  *(bf16x8*)&S[k * TR_L + (xx0 ^ tr_xor(k))] = v;
}

// MFMA fragment (8 reduction elems for column cb + (lane&15)) out of
// the stage_kx image via two transpose-reads.
DEVINL bf16x8 tr_frag(const __bf16* S, int kfrag, int cb, int lane) {
  const int a = (kfrag + ((lane & 15) >> 2)) * TR_L +
                ((cb + 4 * (lane & 3)) ^ tr_xor(kfrag));
  bf16x4 f0 = tr_read(&S[a]);
  bf16x4 f1 = tr_read(&S[a + 4 * TR_L]);
  return __builtin_shufflevector(f0, f1, 0, 1, 2, 3, 4, 5, 6, 7);
}

// ACT: 0 none, 1 relu, 2 relu-bwd (mask by aux>0, the saved activation).
// SK: split-K over gridDim.z with last-arriver epilogue.
// CS: colsum_out[n] = sum_k op(B)[k][n] computed by blockIdx.y==0 WGs.
template <bool TA, bool TB, int ACT, bool BIAS, bool OUTF32, bool SK, bool CS>
__global__ __launch_bounds__(256)
void gemm_kernel(const __bf16* __restrict__ A, const __bf16* __restrict__ B,
                 const void* __restrict__ bias, bool bias_bf16,
                 void* __restrict__ Cout, const __bf16* __restrict__ aux,
                 void* __restrict__ colsum_out, float* __restrict__ ws,
                 int* __restrict__ cnt, int M, int N, int K, int lda, int ldb,
                 int ldc, int kc, int veca, int vecb, int rflags) {
  // double-buffered: stage tile i+1 while MFMA consumes tile i (one
  // barrier per K-iteration). The single-buffered version serialized
  // global-load latency -> barrier -> MFMA every iteration and ran
  // the deep backward shapes at ~1.1 us/iteration (11 TFLOP/s on the
  // NMF dW GEMM).
  // xk-staged operands: [x][k] XOR-swizzled tiles; kx-staged
  // (transposing) operands: reduction-major transpose-read image
  __shared__ __align__(16) __bf16 As[2][TA ? TR_ELEMS : SWZ_ELEMS];
  __shared__ __align__(16) __bf16 Bs[2][TB ? SWZ_ELEMS : TR_ELEMS];

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

#define STAGE_AB(buf, kk0v)                                                 \
  do {                                                                      \
    if (TA) stage_kx(A, As[buf], tm0, kk0v, M, K, lda, t, veca);            \
    else    stage_xk(A, As[buf], tm0, kk0v, M, K, lda, t, veca);            \
    if (TB) stage_xk(B, Bs[buf], tn0, kk0v, N, K, ldb, t, vecb);            \
    else    stage_kx(B, Bs[buf], tn0, kk0v, N, K, ldb, t, vecb);            \
  } while (0)

  STAGE_AB(0, ks);
  __syncthreads();
  int cur = 0;
  for (int k0 = ks; k0 < ke; k0 += BK, cur ^= 1) {
    if (k0 + BK < ke) STAGE_AB(cur ^ 1, k0 + BK);

    if (CS && blockIdx.y == 0 && t < BN) {
#pragma unroll
      for (int kk = 0; kk < BK; ++kk)
        cs_acc += TB ? (float)*sptr(Bs[cur], t, kk)
                     : (float)Bs[cur][kk * TR_L + (t ^ tr_xor(kk))];
    }

    const int kfrag = (lane >> 4) * 8;
    bf16x8 bfrag[2];
#pragma unroll
    for (int fn = 0; fn < 2; ++fn)
      bfrag[fn] = TB
          ? *(const bf16x8*)sptr(Bs[cur],
                                 wc * 32 + fn * 16 + (lane & 15), kfrag)
          : tr_frag(Bs[cur], kfrag, wc * 32 + fn * 16, lane);
#pragma unroll
    for (int fm = 0; fm < 2; ++fm) {
      bf16x8 a = TA
          ? tr_frag(As[cur], kfrag, wr * 32 + fm * 16, lane)
          : *(const bf16x8*)sptr(As[cur],
                                 wr * 32 + fm * 16 + (lane & 15), kfrag);
#pragma unroll
      for (int fn = 0; fn < 2; ++fn)
        acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a, bfrag[fn], acc[fm][fn], 0, 0, 0);
    }
    __syncthreads();
  }
#undef STAGE_AB

  if (CS && blockIdx.y == 0 && t < BN && tn0 + t < N) {
    if (SK)       // per-slice fp32 partial, summed by the reduce kernel
      ((float*)colsum_out)[(long)blockIdx.z * N + tn0 + t] = cs_acc;
    else if (OUTF32) ((float*)colsum_out)[tn0 + t] = cs_acc;
    else ((__bf16*)colsum_out)[tn0 + t] = (__bf16)cs_acc;
  }

  if (SK) {
    // store this slice's fp32 partial tile into its workspace stripe;
    // the follow-up reduce kernel sums stripes in fixed slice order —
    // or, when cnt != nullptr (small tile grids), the LAST-ARRIVING
    // slice block sums them here and applies the epilogue itself
    // (fixed z order, so still deterministic), saving the reduce
    // launch + its dependent-kernel boundary. rflags: bit0 relu,
    // bit1 fp32 out, bit2 bias.
    float* wslice = ws + (long)blockIdx.z * (long)M * ldc;
#pragma unroll
    for (int fm = 0; fm < 2; ++fm)
#pragma unroll
      for (int fn = 0; fn < 2; ++fn) {
        const int col = tn0 + wc * 32 + fn * 16 + (lane & 15);
        if (col >= N) continue;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int row = tm0 + wr * 32 + fm * 16 + (lane >> 4) * 4 + r;
          if (row < M) wslice[(long)row * ldc + col] = acc[fm][fn][r];
        }
      }
    if (cnt != nullptr) {
      __threadfence();
      __shared__ int lastf;
      if (t == 0)
        lastf = (atomicAdd(&cnt[blockIdx.y * gridDim.x + blockIdx.x], 1) ==
                 (int)gridDim.z - 1) ? 1 : 0;
      __syncthreads();
      if (!lastf) return;
      __threadfence();   // acquire: don't serve stale L2 for the stripes
      if (t == 0) cnt[blockIdx.y * gridDim.x + blockIdx.x] = 0;
      const int ns = gridDim.z;
      // vectorized row-major sweep of the whole 64x64 tile: 4 f32x4
      // positions per thread, every stripe's loads independent — the
      // first cut walked the MFMA acc mapping with scalar loads and
      // the lone consumer block serialized ~144 cross-XCD round trips
      // (the one-WG-consumer trap, see mlp.py's FROMWS note)
      const int crow = t >> 2;             // 0..63 within tile
      const int ccol0 = (t & 3) * 16;      // 4 f32x4 groups per row
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        const int row = tm0 + crow;
        const int col = tn0 + ccol0 + u * 4;
        if (row >= M || col >= N) continue;
        const long idx = (long)row * ldc + col;
        f32x4 v = {};
        for (int z2 = 0; z2 < ns; ++z2) {
          const float* s2 = &ws[(long)z2 * M * ldc + idx];
          if (col + 4 <= N) {
            v += *(const f32x4*)s2;
          } else {
            for (int j = 0; j < 4; ++j)
              if (col + j < N) v[j] += s2[j];
          }
        }
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          if (col + j >= N) continue;
          float x = v[j];
          if (rflags & 4)
            x += bias_bf16 ? (float)((const __bf16*)bias)[col + j]
                           : ((const float*)bias)[col + j];
          if (rflags & 1) x = x > 0.f ? x : 0.f;
          if (rflags & 2) ((float*)Cout)[idx + j] = x;
          else ((__bf16*)Cout)[idx + j] = (__bf16)x;
        }
      }
    }
    return;
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
        float v = acc[fm][fn][r];
        v += bv;
        if (ACT == 1) v = v > 0.f ? v : 0.f;
        if (ACT == 2) v = (float)aux[idx] > 0.f ? v : 0.f;
        if (ACT == 3) v -= (float)aux[idx];   // fused residual (NMF E)
        if (OUTF32)
          ((float*)Cout)[idx] = v;
        else
          ((__bf16*)Cout)[idx] = (__bf16)v;
      }
    }
  }
}

// split-K phase 2: out[i] = act(sum_z ws[z][i] + bias), f32x4-vectorized
// over the flattened [M,ldc] output (ldc == N, contiguous).
// 1-elem/thread variant for SMALL outputs (mnist fwd: mn = 10k gave
// only 10 WGs at 4 elems/thread — the reduce was WG-count
// latency-bound, not bandwidth-bound; 4x the blocks cut it ~2x)
template <int ACT, bool BIAS, bool OUTF32>
__global__ __launch_bounds__(256)
void splitk_reduce_small_kernel(const float* __restrict__ ws,
                                const void* __restrict__ bias,
                                bool bias_bf16, void* __restrict__ Cout,
                                long mn, int ldc, int nslice) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= mn) return;
  float v = 0.f;
  for (int z = 0; z < nslice; ++z) v += ws[(long)z * mn + i];
  if (BIAS) {
    const int col = (int)(i % ldc);
    v += bias_bf16 ? (float)((const __bf16*)bias)[col]
                   : ((const float*)bias)[col];
  }
  if (ACT == 1) v = v > 0.f ? v : 0.f;
  if (OUTF32) ((float*)Cout)[i] = v;
  else ((__bf16*)Cout)[i] = (__bf16)v;
}

template <int ACT, bool BIAS, bool OUTF32>
__global__ __launch_bounds__(256)
void splitk_reduce_kernel(const float* __restrict__ ws, const void* __restrict__ bias,
                          bool bias_bf16, void* __restrict__ Cout, long mn,
                          int ldc, int nslice,
                          const float* __restrict__ cs_part,
                          void* __restrict__ cs_out, bool cs_f32, int N) {
  const long i0 = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 4;
  if (i0 >= mn) return;
  if (cs_part != nullptr && i0 < N) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const long n = i0 + j;
      if (n >= N) break;
      float s = 0.f;
      for (int z = 0; z < nslice; ++z) s += cs_part[(long)z * N + n];
      if (cs_f32) ((float*)cs_out)[n] = s;
      else ((__bf16*)cs_out)[n] = (__bf16)s;
    }
  }
  f32x4 v = {};
  for (int z = 0; z < nslice; ++z) {
    const float* s = ws + (long)z * mn + i0;
    if (i0 + 4 <= mn) {
      const f32x4 sv = *(const f32x4*)s;
#pragma unroll
      for (int j = 0; j < 4; ++j) v[j] += sv[j];
    } else {
      for (int j = 0; i0 + j < mn; ++j) v[j] += s[j];
    }
  }
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    const long i = i0 + j;
    if (i >= mn) break;
    float x = v[j];
    if (BIAS) {
      const int col = (int)(i % ldc);
      x += bias_bf16 ? (float)((const __bf16*)bias)[col]
                     : ((const float*)bias)[col];
    }
    if (ACT == 1) x = x > 0.f ? x : 0.f;
    if (OUTF32) ((float*)Cout)[i] = x;
    else ((__bf16*)Cout)[i] = (__bf16)x;
  }
}


typedef __attribute__((ext_vector_type(16))) float f32x16;

// ---------------------------------------------------------- small-tile path
//
// One-kernel GEMM for tiny tile grids (the mnist fwd 100x100x784 and
// dW1 784x100x100): the two-phase split-K path paid TWO dispatch
// floors (~13 us for ~16 MFLOP). 32x32 output tile per block; the 4
// waves SPLIT K privately — each stages into its OWN LDS quarter, so
// the k-loop has NO barrier — then one barrier, a cross-wave
// accumulate by wave 0, and the fused epilogue (bias/relu, bf16/f32
// out). colsum (db1) is summed during B staging and reduced by
// blockIdx.x==0 only (it depends on K alone, so every m-tile block
// would compute the same value; single-writer keeps it deterministic,
// no atomics, overwrite semantics like the 64x64 path).
// TA: op(A) = A^T (A stored [K,M] — transposed scatter staging).
// B is always untransposed here ([K,N] memory).
//
// SmallSgd: optional fused optimizer tail (the mnist single-GPU fast
// path). When pmw != null the epilogue APPLIES this GEMM's output as
// an SGD gradient (master -= lr*g, refresh bf16 shadow) instead of
// storing it, the colsum applies the bias grad the same way, and one
// otherwise-idle wave of block (0,0) also applies the classifier
// grads (g2w/g2b, written by the head kernel that ran just before) —
// the step's separate flat-buffer sgd_kernel launch disappears.
// Valid only for plain SGD (no momentum/decay) with grad_scale 1
// (single worker); the distributed path keeps the PS apply.
struct SmallSgd {
  float* pmw;          // W master [M,N] fp32 (same layout as Cout)
  bf16_t* psw;         // W bf16 shadow
  float* pmb;          // bias master [N]
  bf16_t* psb;
  const bf16_t* g2w;   // classifier grads (from the head kernel)
  float* pm2w;
  bf16_t* ps2w;
  const bf16_t* g2b;
  float* pm2b;
  bf16_t* ps2b;
  float lr;
  int n2w, n2b;
};

template <bool TA, bool CS>
__global__ __launch_bounds__(256)
void gemm_small_kernel(const __bf16* __restrict__ A,
                       const __bf16* __restrict__ B,
                       const void* __restrict__ bias, bool bias_bf16,
                       void* __restrict__ Cout, bool out_f32, int relu,
                       void* __restrict__ colsum_out, bool cs_f32,
                       int M, int N, int K, int lda, int ldb, int ldc,
                       SmallSgd sg) {
  // wave-private 32x32 images, stride exactly 32 + the sptr chunk
  // swizzle: the transposing operands (A when TA, B always) store
  // K-MAJOR [k][col] — commit is two b128 stores (the transposed
  // scatter commit measured 3.5 extra LDS-conflict cycles per LDS
  // instruction here) — and fragments come back via ds_read_b64_tr_b16
  __shared__ __align__(16) __bf16 As[4][32 * BK];
  __shared__ __align__(16) __bf16 Bs[4][32 * BK];
  __shared__ float red[4][64 * 16];
  __shared__ float csred[4][32];
  const int t = threadIdx.x;
  const int lane = t & 63;
  const int w = t >> 6;
  const int m0 = blockIdx.x * 32;
  const int n0 = blockIdx.y * 32;
  const int kq = (((K + 3) / 4) + BK - 1) / BK * BK;  // per-wave K span
  const int ks = w * kq;
  const int ke = min(ks + kq, K);
  __bf16* Aw = As[w];
  __bf16* Bw = Bs[w];
  const int kk = lane & 31;
  const int half = lane >> 5;
  const bool av = (lda & 7) == 0;   // bf16x8-aligned rows
  const bool bv = (ldb & 7) == 0;
  f32x4 acc[4] = {};
  float csp[16] = {};
  bf16x8 ra0, ra1, rb0, rb1;        // in-flight chunk (load -> commit)

  // load: global reads into REGISTERS (vector when the row allows —
  // guarded scalar loads serialized the first cut at 2x the split-K
  // path); commit: LDS stores. Registers double-buffer the single LDS
  // image: the per-wave DS pipe is in-order, so this k-chunk's
  // fragment reads are serviced before the next chunk's commit writes.
  auto load_a = [&](int k0) {
    ra0 = bf16x8{};
    ra1 = bf16x8{};
    if (TA) {
      // A stored [K, M]: 16 consecutive m of one k
      const int gk = k0 + kk;
      const int gm = m0 + half * 16;
      if (gk < ke) {
        const __bf16* src = A + (long)gk * lda + gm;
        if (av && gm + 16 <= M && ((m0 & 7) == 0)) {
          ra0 = *(const bf16x8*)src;
          ra1 = *(const bf16x8*)(src + 8);
        } else {
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            if (gm + j < M) ra0[j] = src[j];
            if (gm + 8 + j < M) ra1[j] = src[8 + j];
          }
        }
      }
    } else {
      // A stored [M, K]: row m, 16 consecutive k
      const int gm = m0 + kk;
      const int gk0 = k0 + half * 16;
      if (gm < M) {
        const __bf16* src = A + (long)gm * lda + gk0;
        if (av && gk0 + 16 <= ke) {
          ra0 = *(const bf16x8*)src;
          ra1 = *(const bf16x8*)(src + 8);
        } else {
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            if (gk0 + j < ke) ra0[j] = src[j];
            if (gk0 + 8 + j < ke) ra1[j] = src[8 + j];
          }
        }
      }
    }
    // B stored [K, N]: 16 consecutive n of one k
    rb0 = bf16x8{};
    rb1 = bf16x8{};
    {
      const int gk = k0 + kk;
      const int gn = n0 + half * 16;
      if (gk < ke) {
        const __bf16* src = B + (long)gk * ldb + gn;
        if (bv && gn + 16 <= N && ((n0 & 7) == 0)) {
          rb0 = *(const bf16x8*)src;
          rb1 = *(const bf16x8*)(src + 8);
        } else {
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            if (gn + j < N) rb0[j] = src[j];
            if (gn + 8 + j < N) rb1[j] = src[8 + j];
          }
        }
      }
    }
  };
  auto commit = [&]() {
    // both layouts have row kk: TA-A/B rows are k (16 consecutive
    // cols of one k per register), non-TA A rows are m (16
    // consecutive k) — either way two b128 chunk-aligned stores
    *(bf16x8*)sptr(Aw, kk, half * 16) = ra0;
    *(bf16x8*)sptr(Aw, kk, half * 16 + 8) = ra1;
    *(bf16x8*)sptr(Bw, kk, half * 16) = rb0;
    *(bf16x8*)sptr(Bw, kk, half * 16 + 8) = rb1;
    if (CS) {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        csp[j] += (float)rb0[j];
        csp[8 + j] += (float)rb1[j];
      }
    }
  };

  // transpose-read fragment out of a k-major wave image (see tr_frag;
  // same provider-lane scheme, 32-wide rows via sptr)
  auto trs_frag = [&](const __bf16* S, int ko) {
    const int kq = ko + ((lane & 15) >> 2);
    const int c = 16 * ((lane >> 4) & 1) + 4 * (lane & 3);
    bf16x4 f0 = tr_read(sptr((__bf16*)S, kq, c));
    bf16x4 f1 = tr_read(sptr((__bf16*)S, kq + 4, c));
    return __builtin_shufflevector(f0, f1, 0, 1, 2, 3, 4, 5, 6, 7);
  };

  load_a(ks);
  commit();
  for (int k0 = ks; k0 < ke; k0 += BK) {
    const bool more = k0 + BK < ke;
    if (more) load_a(k0 + BK);
#pragma unroll
    for (int kh = 0; kh < 2; ++kh) {
      const int ko = kh * 16 + ((lane >> 5) << 3);
      bf16x8 a = TA ? trs_frag(Aw, ko)
                    : *(const bf16x8*)sptr(Aw, lane & 31, ko);
      bf16x8 b = trs_frag(Bw, ko);
      *(f32x16*)acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
          a, b, *(f32x16*)acc, 0, 0, 0);
    }
    if (more) commit();
  }

  // cross-wave reduce (the only barrier) + epilogue by wave 0
#pragma unroll
  for (int v = 0; v < 16; ++v) red[w][lane * 16 + v] = ((float*)acc)[v];
  if (CS && blockIdx.x == 0) {
    float c2[16];
#pragma unroll
    for (int j = 0; j < 16; ++j) c2[j] = csp[j];
#pragma unroll
    for (int off = 1; off < 32; off <<= 1)
#pragma unroll
      for (int j = 0; j < 16; ++j) c2[j] += __shfl_xor(c2[j], off, 64);
    if (kk == 0) {
#pragma unroll
      for (int j = 0; j < 16; ++j) csred[w][half * 16 + j] = c2[j];
    }
  }
  __syncthreads();
  if (sg.pm2w != nullptr && w == 1 && blockIdx.x == 0 && blockIdx.y == 0) {
    // classifier apply: ~1k elems on one idle wave; runs strictly
    // after the head kernel wrote g2w/g2b (stream order), and nothing
    // reads the classifier shadow until the next step's head.
    // Vectorized 4-deep with every load issued before any store — the
    // first cut (scalar strided loop) serialized ~16 global round
    // trips on this single wave and cost MORE than the sgd_kernel it
    // replaced. Buffers are 256-elem-aligned flat-store slices, so
    // the f32x4/bf16x4 accesses are aligned.
    const int nv = sg.n2w >> 2;
    f32x4 mv[4];
    bf16x4 gv[4];
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      const int i = lane + u * 64;
      if (i < nv) {
        gv[u] = ((const bf16x4*)sg.g2w)[i];
        mv[u] = ((const f32x4*)sg.pm2w)[i];
      }
    }
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      const int i = lane + u * 64;
      if (i < nv) {
        f32x4 pv;
        bf16x4 sv;
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          pv[j] = mv[u][j] - sg.lr * (float)gv[u][j];
          sv[j] = (__bf16)pv[j];
        }
        ((f32x4*)sg.pm2w)[i] = pv;
        ((bf16x4*)sg.ps2w)[i] = sv;
      }
    }
    for (int i = 256 + lane; i < nv; i += 64) {   // >64 KB W2 (not mnist)
      f32x4 pv;
      bf16x4 sv;
      const bf16x4 g = ((const bf16x4*)sg.g2w)[i];
      const f32x4 m = ((const f32x4*)sg.pm2w)[i];
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        pv[j] = m[j] - sg.lr * (float)g[j];
        sv[j] = (__bf16)pv[j];
      }
      ((f32x4*)sg.pm2w)[i] = pv;
      ((bf16x4*)sg.ps2w)[i] = sv;
    }
    for (int i = (nv << 2) + lane; i < sg.n2w; i += 64) {   // n2w % 4
      const float pv = sg.pm2w[i] - sg.lr * bf2f(sg.g2w[i]);
      sg.pm2w[i] = pv;
      sg.ps2w[i] = f2bf(pv);
    }
    for (int i = lane; i < sg.n2b; i += 64) {
      const float pv = sg.pm2b[i] - sg.lr * bf2f(sg.g2b[i]);
      sg.pm2b[i] = pv;
      sg.ps2b[i] = f2bf(pv);
    }
  }
  if (sg.pmw != nullptr) {
    // fused W1/b1 apply: ALL FOUR waves split the 16 output values
    // (4 each, straight from the LDS partials) — the wave0-only form
    // serialized the master-read/store latency on one wave per block
    // and measured SLOWER than the separate wide sgd_kernel
    const int n = n0 + (lane & 31);
    if (n < N) {
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        const int v = w * 4 + u;
        const int m = m0 + ((v >> 2) << 3) + ((lane >> 5) << 2) + (v & 3);
        if (m >= M) continue;
        const float x = red[0][lane * 16 + v] + red[1][lane * 16 + v] +
                        red[2][lane * 16 + v] + red[3][lane * 16 + v];
        const long idx = (long)m * ldc + n;
        const float pv = sg.pmw[idx] - sg.lr * x;
        sg.pmw[idx] = pv;
        sg.psw[idx] = f2bf(pv);
      }
    }
    if (CS && w == 0 && blockIdx.x == 0 && lane < 32) {
      const int nn = n0 + lane;
      if (nn < N && sg.pmb != nullptr) {
        const float sv = csred[0][lane] + csred[1][lane] + csred[2][lane] +
                         csred[3][lane];
        const float pv = sg.pmb[nn] - sg.lr * sv;
        sg.pmb[nn] = pv;
        sg.psb[nn] = f2bf(pv);
      }
    }
    return;
  }
  if (w != 0) return;
  f32x16 tot;
#pragma unroll
  for (int v = 0; v < 16; ++v)
    tot[v] = red[0][lane * 16 + v] + red[1][lane * 16 + v] +
             red[2][lane * 16 + v] + red[3][lane * 16 + v];
  const int n = n0 + (lane & 31);
  if (n < N) {
    float bvv = 0.f;
    if (bias != nullptr)
      bvv = bias_bf16 ? (float)((const __bf16*)bias)[n]
                      : ((const float*)bias)[n];
#pragma unroll
    for (int v = 0; v < 16; ++v) {
      const int m = m0 + ((v >> 2) << 3) + ((lane >> 5) << 2) + (v & 3);
      if (m >= M) continue;
      float x = tot[v] + bvv;
      if (relu) x = x > 0.f ? x : 0.f;
      if (out_f32) ((float*)Cout)[(long)m * ldc + n] = x;
      else ((__bf16*)Cout)[(long)m * ldc + n] = (__bf16)x;
    }
  }
  if (CS && blockIdx.x == 0 && lane < 32) {
    const int nn = n0 + lane;
    if (nn < N) {
      const float sv = csred[0][lane] + csred[1][lane] + csred[2][lane] +
                       csred[3][lane];
      if (cs_f32) ((float*)colsum_out)[nn] = sv;
      else ((__bf16*)colsum_out)[nn] = (__bf16)sv;
    }
  }
}


// split-K phase 2 fused with the SGD apply (the NMF factor update):
// instead of materializing the fp32 gradient and re-reading it in a
// separate apply kernel, the stripe sum feeds p -= lr*(gscale*g +
// nd*min(p,0)) and the bf16 shadow refresh directly.
__global__ __launch_bounds__(256)
void splitk_reduce_sgd_kernel(const float* __restrict__ ws, int nslice,
                              float* __restrict__ p,
                              __bf16* __restrict__ shadow, long mn,
                              float lr, float gscale, float nd) {
  const long i0 = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 4;
  if (i0 >= mn) return;
  f32x4 v = {};
  for (int z = 0; z < nslice; ++z) {
    const float* sp = ws + (long)z * mn + i0;
    if (i0 + 4 <= mn) {
      const f32x4 sv = *(const f32x4*)sp;
#pragma unroll
      for (int j = 0; j < 4; ++j) v[j] += sv[j];
    } else {
      for (int j = 0; i0 + j < mn; ++j) v[j] += sp[j];
    }
  }
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    const long i = i0 + j;
    if (i >= mn) break;
    float pv = p[i];
    float g = gscale * v[j];
    if (nd != 0.f) g += nd * fminf(pv, 0.f);
    pv -= lr * g;
    p[i] = pv;
    if (shadow != nullptr) shadow[i] = (__bf16)pv;
  }
}

}  // namespace

void launch_gemm(const bf16_t* A, const bf16_t* B, const void* bias,
                 bool bias_bf16, void* C, bool out_f32, const bf16_t* aux,
                 void* colsum_out, float* ws, int* cnt, int kc, int nslice,
                 int M, int N, int K, int lda, int ldb, int ldc, bool ta,
                 bool tb, int act, int veca, int vecb, hipStream_t stream) {
  const bool sk = nslice > 1;
  dim3 grid(ceil_div(N, BN), ceil_div(M, BM), sk ? nslice : 1);
  dim3 block(256);
  const bool has_bias = bias != nullptr;
  const bool cs = colsum_out != nullptr;
  const int rflags = (act == 1 ? 1 : 0) | (out_f32 ? 2 : 0) |
                     (has_bias ? 4 : 0);
  // last-arriver epilogue: MEASURED DEAD END on the mnist fwd shape
  // (4 tiles x 9 slices): the 4 consumer blocks pull the stripes
  // cross-XCD through the release/acquire fences and the fwd GEMM
  // went 8.2 -> 48 us scalar / ~20 us vectorized, vs 12.9 us for
  // stripes + the wide reduce kernel. Same lesson as the FROMWS head:
  // few-WG consumers must not read many-XCD-producer output. Kept
  // behind TFA_GEMM_LA=1 for re-measurement; default off.
  static int la_on = -1;
  if (la_on < 0) {
    const char* e = getenv("TFA_GEMM_LA");
    la_on = e ? atoi(e) : 0;
  }
  if (!la_on || cs || act > 1 ||
      (long)ceil_div(N, BN) * ceil_div(M, BM) > 64)
    cnt = nullptr;
#define LAUNCH(TAv, TBv, ACTv, BIASv, OUTv, SKv, CSv)                       \
  hipLaunchKernelGGL((gemm_kernel<TAv, TBv, ACTv, BIASv, OUTv, SKv, CSv>),  \
                     grid, block, 0, stream, (const __bf16*)A,              \
                     (const __bf16*)B, bias, bias_bf16, C,                  \
                     (const __bf16*)aux, colsum_out, ws, cnt, M, N, K, lda, \
                     ldb, ldc, kc, veca, vecb, rflags)
  // constraint combos (relu_bwd only nt/bf16, colsum only tn) are
  // validated by the bindings in ext.hip
  if (act == 2) {
    LAUNCH(false, true, 2, false, false, false, false);
    return;
  }
  if (act == 3) {   // out = A@B - aux (nn, bf16 out; binding-gated)
    LAUNCH(false, false, 3, false, false, false, false);
    return;
  }
  if (cs && !sk) {
    if (out_f32) LAUNCH(true, false, 0, false, true, false, true);
    else         LAUNCH(true, false, 0, false, false, false, true);
    return;
  }
  if (cs && sk) {
    // phase 1: tn with per-slice output AND colsum partials; phase 2
    // reduces both. cs_part lives at ws + mn*nslice (sized by binding)
    const long mn = (long)M * ldc;
    float* cs_part = ws + mn * nslice;
    void* user_cs = colsum_out;
    colsum_out = cs_part;    // phase 1 writes the partials
    LAUNCH(true, false, 0, false, false, true, true);
    colsum_out = user_cs;
    dim3 rgrid((unsigned)(((mn + 3) / 4 + 255) / 256)), rblock(256);
    if (out_f32)
      hipLaunchKernelGGL((splitk_reduce_kernel<0, false, true>), rgrid,
                         rblock, 0, stream, ws, nullptr, false, C, mn, ldc,
                         nslice, cs_part, colsum_out, out_f32, N);
    else
      hipLaunchKernelGGL((splitk_reduce_kernel<0, false, false>), rgrid,
                         rblock, 0, stream, ws, nullptr, false, C, mn, ldc,
                         nslice, cs_part, colsum_out, out_f32, N);
    return;
  }
  if (sk) {
    // phase 1: partials (epilogue template args unused in SK stores);
    // all four transpose combos (deep-K BACKWARD shapes with small
    // tile grids — e.g. NMF dW [1000,200,k=1000] — were 1 wave/SIMD
    // and LDS-latency-bound without slicing)
    if (ta) { if (tb) LAUNCH(true, true, 0, false, false, true, false);
              else    LAUNCH(true, false, 0, false, false, true, false); }
    else    { if (tb) LAUNCH(false, true, 0, false, false, true, false);
              else    LAUNCH(false, false, 0, false, false, true, false); }
    if (cnt != nullptr) return;   // last-arriver epilogue ran in-kernel
    // phase 2: fixed-order stripe reduce + fused epilogue (small
    // outputs: 1 elem/thread for 4x the workgroups)
    const long mn = (long)M * ldc;
    const bool small_r = mn <= (1 << 16);
    dim3 rgrid((unsigned)(small_r ? (mn + 255) / 256
                                  : ((mn + 3) / 4 + 255) / 256));
    dim3 rblock(256);
#define RLAUNCH(ACTv, BIASv, OUTv)                                          \
    do { if (small_r)                                                       \
      hipLaunchKernelGGL((splitk_reduce_small_kernel<ACTv, BIASv, OUTv>),   \
                         rgrid, rblock, 0, stream, ws, bias, bias_bf16, C,  \
                         mn, ldc, nslice);                                  \
    else                                                                    \
      hipLaunchKernelGGL((splitk_reduce_kernel<ACTv, BIASv, OUTv>), rgrid,  \
                         rblock, 0, stream, ws, bias, bias_bf16, C, mn,     \
                         ldc, nslice, (const float*)nullptr,                \
                         (void*)nullptr, false, 0);                         \
    } while (0)
    if (act == 1) {
      if (has_bias) { if (out_f32) RLAUNCH(1, true, true);
                      else RLAUNCH(1, true, false); }
      else          { if (out_f32) RLAUNCH(1, false, true);
                      else RLAUNCH(1, false, false); }
    } else {
      if (has_bias) { if (out_f32) RLAUNCH(0, true, true);
                      else RLAUNCH(0, true, false); }
      else          { if (out_f32) RLAUNCH(0, false, true);
                      else RLAUNCH(0, false, false); }
    }
#undef RLAUNCH
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

// split-K phase 1 ONLY: store the fp32 partial stripes and return —
// the caller fuses the stripe reduction into a consumer kernel (the
// mnist fused head reduces them while staging h into LDS, so h never
// exists in global memory and the reduce launch disappears).

void launch_gemm_small(const bf16_t* A, const bf16_t* B, const void* bias,
                       bool bias_bf16, void* C, bool out_f32, int relu,
                       void* colsum_out, bool cs_f32, int M, int N, int K,
                       int lda, int ldb, int ldc, bool ta,
                       const SmallSgdArgs* sga, hipStream_t stream) {
  dim3 grid(ceil_div(M, 32), ceil_div(N, 32)), block(256);
  SmallSgd sg = {};
  if (sga) {
    sg.pmw = sga->pmw; sg.psw = (bf16_t*)sga->psw;
    sg.pmb = sga->pmb; sg.psb = (bf16_t*)sga->psb;
    sg.g2w = (const bf16_t*)sga->g2w;
    sg.pm2w = sga->pm2w; sg.ps2w = (bf16_t*)sga->ps2w;
    sg.g2b = (const bf16_t*)sga->g2b;
    sg.pm2b = sga->pm2b; sg.ps2b = (bf16_t*)sga->ps2b;
    sg.lr = sga->lr; sg.n2w = sga->n2w; sg.n2b = sga->n2b;
  }
  const bool cs = colsum_out != nullptr || sg.pmb != nullptr;
#define GS(TAv, CSv)                                                        \
  hipLaunchKernelGGL((gemm_small_kernel<TAv, CSv>), grid, block, 0,         \
                     stream, (const __bf16*)A, (const __bf16*)B, bias,      \
                     bias_bf16, C, out_f32, relu, colsum_out, cs_f32, M, N, \
                     K, lda, ldb, ldc, sg)
  if (ta) { if (cs) GS(true, true); else GS(true, false); }
  else    { if (cs) GS(false, true); else GS(false, false); }
#undef GS
}


// phase-1-only split-K stripes for ANY transpose combo (the fused
// GEMM->SGD path reduces them with splitk_reduce_sgd_kernel)
void launch_gemm_stripes_any(const bf16_t* A, const bf16_t* B, float* ws,
                             int kc, int nslice, int M, int N, int K,
                             int lda, int ldb, int ldc, bool ta, bool tb,
                             int veca, int vecb, hipStream_t stream) {
  const long Mtiles = ((long)M + BM - 1) / BM;
  dim3 grid((unsigned)ceil_div(N, BN), (unsigned)Mtiles, nslice);
  dim3 block(256);
#define SLAUNCH(TAv, TBv)                                                   \
  hipLaunchKernelGGL((gemm_kernel<TAv, TBv, 0, false, false, true, false>), \
                     grid, block, 0, stream, (const __bf16*)A,              \
                     (const __bf16*)B, nullptr, false, nullptr, nullptr,    \
                     nullptr, ws, nullptr, M, N, K, lda, ldb, ldc, kc, veca,\
                     vecb, 0)
  if (ta) { if (tb) SLAUNCH(true, true); else SLAUNCH(true, false); }
  else    { if (tb) SLAUNCH(false, true); else SLAUNCH(false, false); }
#undef SLAUNCH
}

void launch_splitk_reduce_sgd(const float* ws, int nslice, float* p,
                              bf16_t* shadow, long mn, float lr,
                              float gscale, float nd, hipStream_t stream) {
  dim3 rgrid((unsigned)(((mn + 3) / 4 + 255) / 256)), rblock(256);
  hipLaunchKernelGGL(splitk_reduce_sgd_kernel, rgrid, rblock, 0, stream,
                     ws, nslice, p, (__bf16*)shadow, mn, lr, gscale, nd);
}

void launch_gemm_stripes(const bf16_t* A, const bf16_t* B, float* ws,
                         int kc, int nslice, int M, int N, int K, int lda,
                         int ldb, int ldc, int veca, int vecb,
                         hipStream_t stream) {
  dim3 grid(ceil_div(N, BN), ceil_div(M, BM), nslice);
  dim3 block(256);
  hipLaunchKernelGGL((gemm_kernel<false, false, 0, false, false, true,
                                  false>),
                     grid, block, 0, stream, (const __bf16*)A,
                     (const __bf16*)B, nullptr, false, nullptr, nullptr,
                     nullptr, ws, nullptr, M, N, K, lda, ldb, ldc, kc,
                     veca, vecb, 0);
}
