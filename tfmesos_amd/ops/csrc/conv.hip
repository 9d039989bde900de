// Implicit-GEMM 2-D convolution on MFMA, channels-last (NHWC), bf16 in,
// fp32 accumulate.
//
// The Inception-v3 config of BASELINE.json needs conv fwd/bwd on the
// worker compute path (the reference delegated convs to TF kernels;
// SURVEY.md §2b row "conv stack"). CDNA4-native design: the conv is a
// GEMM over [output pixels] x [filter taps] with the im2col gather fused
// into LDS staging (no materialized col buffer), the same
// v_mfma_f32_16x16x32_bf16 64x64 core as gemm.hip, and CHANNELS-LAST
// memory so every gather is a contiguous run of C (the profile of the
// NCHW version, profiles/r01_inception_n1_kernel_stats.txt, showed the
// channel-strided scalar gathers capping the kernels at ~1.5% MFMA
// peak). Tap order is (r, s, c) — c innermost — so an 8-element strip
// is ONE aligned b128 load whenever C % 8 == 0 (every Inception layer
// except the 3-channel stem input; scalar fallback otherwise), and the
// output store is coalesced across channels.
//
//   fwd:    Y[n,ho,wo,k]  = sum_{r,s,c} X[n, ho*U+r-P, wo*V+s-Q, c] W[k,r,s,c]
//           GEMM  M = N*Ho*Wo, Ncol = K,     Kdim = R*S*C
//   bwd-d:  dX[n,h,w,c]   = sum_{r,s,k} dY[n,(h+P-r)/U,(w+Q-s)/V,k] W[k,r,s,c]
//           GEMM  M = N*H*W,   Ncol = C,     Kdim = R*S*K  (W read in its
//           native [k][r][s][c] layout by a transposing stager)
//   bwd-w:  dW[k,r,s,c]   = sum_{n,ho,wo} dY[n,ho,wo,k] X[n,ho*U+r-P,...,c]
//           GEMM  M = K,       Ncol = R*S*C, Kdim = N*Ho*Wo
#include <cstdlib>
#include <type_traits>

#include "common.h"

namespace {

constexpr int BM = 64, BN = 64, BK = 32;
// XOR-swizzled [64 rows][32 cols] bf16 tile, row stride EXACTLY 32
// elems: elem (r, c) lives at r*32 + (((c>>3) ^ ((r>>2)&3))*8 + (c&7)).
// ds_read_b128 is serviced in four NON-CONTIGUOUS 16-lane groups
// ({0-3,12-15,20-27} etc. — MI355X_MICROARCH.md §LDS), and the earlier
// pad+skew row stride (40 elems + 16 B per 8 rows) still left 2-way
// bank aliases inside those groups (measured: SQ_LDS_BANK_CONFLICT ~
// 3-6x e9 over a convbench pass on the fwd/bwd-data kernels). With
// stride 32 a row's quarter of the banks is r&3; XOR-ing the 8-elem
// CHUNK index by (r>>2)&3 makes (quarter, chunk-slot) injective over
// every b128 16-lane group pattern AND keeps the b128 staging writes
// (contiguous 8-lane groups, bank mod 32) conflict-free. All accesses
// are 8-chunk-aligned or per-element, so the swizzle never splits a
// vector access. Bonus: 2048-elem tiles (was 2624).
constexpr int TILE_ELEMS = BM * BK;

typedef __attribute__((ext_vector_type(16))) float f32x16;

DEVINL __bf16* sptr(__bf16* S, int r, int c) {
  return S + r * BK + ((((c >> 3) ^ ((r >> 2) & 3)) << 3) | (c & 7));
}
DEVINL const __bf16* sptr(const __bf16* S, int r, int c) {
  return S + r * BK + ((((c >> 3) ^ ((r >> 2) & 3)) << 3) | (c & 7));
}

// Magic-number unsigned division (Granlund-Montgomery): integer divide
// on CDNA is ~30 VALU cycles and the gathers decode several indices per
// 8-strip. With L = ceil(log2 d) and M = ceil(2^(31+L) / d),
// floor(x/d) == (x * M) >> (31+L) exactly for all 0 <= x < 2^31.
struct FastDiv {
  unsigned long long m;
  int s;                 // 31 + L
  int d;
};

inline FastDiv make_fdiv(int d) {
  FastDiv f;
  f.d = d;
  int L = 0;
  while ((1LL << L) < d) ++L;
  f.s = 31 + L;
  f.m = ((1ULL << f.s) + d - 1) / d;
  return f;
}

DEVINL unsigned fdiv(unsigned x, FastDiv f) {
  return (unsigned)(((unsigned long long)x * f.m) >> f.s);
}

struct ConvShape {
  int N, C, H, W;     // input (NHWC memory)
  int K, R, S;        // filter
  int Ho, Wo;         // output
  int U, V;           // stride
  int P, Q;           // pad
  FastDiv dC, dS, dK, dWo, dW2, dHoWo, dHW;   // divisors for gather decode
  int LDY;   // dY row stride (== K, or wider when dY is a channel-
             // narrow view of a concat-grad buffer — no contiguous copy)
};

// ---------------------------------------------------------------- forward

// As[m][kk]: m = output pixel, kk = tap (r,s,c) with c innermost.
// Thread t: pixel row m = t>>2, 8-tap chunk kk0 = (t&3)*8 -> one b128.
// The pixel decode (n,ho,wo) is FIXED per thread for the whole tap
// reduction (the m-tile never moves), so it is hoisted into init();
// only the tap decode runs per tile — 2 fdivs instead of 4.
struct FwdPatchStage {
  int mx, kk0;
  int n, ho, wo;
  bool pok;

  DEVINL void init(const ConvShape& cs, long m0, long M, int t) {
    mx = t >> 2;
    kk0 = (t & 3) * 8;
    const long pm = m0 + mx;
    pok = pm < M;
    const unsigned pc = (unsigned)(pok ? pm : 0);
    const int HoWo = cs.Ho * cs.Wo;
    n = (int)fdiv(pc, cs.dHoWo);
    const int rem = (int)(pc - (unsigned)n * HoWo);
    ho = fdiv(rem, cs.dWo);
    wo = rem - ho * cs.Wo;
  }

  bf16x8 vv;   // in-flight tile (load() -> commit())

  // split load/commit: load() issues the global reads into registers
  // BEFORE the MFMA block (their latency overlaps compute); commit()
  // is one b128 LDS store after the MFMAs. The fused form parked each
  // wave on vmcnt ahead of its own MFMAs.
  DEVINL void load(const __bf16* __restrict__ X, const ConvShape cs,
                   int k0, int KD, bool cvec) {
    const int q0 = k0 + kk0;
    vv = {};
    if (q0 >= KD || !pok) return;
    const int rs = fdiv(q0, cs.dC);    // tap (r,s) block (c0 % 8 == 0 when
    const int c0 = q0 - rs * cs.C;     //  C % 8 == 0, so the run stays
    const int r = fdiv(rs, cs.dS);     //  inside one (r,s))
    const int s = rs - r * cs.S;
    const int hi = ho * cs.U + r - cs.P;
    const int wi = wo * cs.V + s - cs.Q;

    if (cvec && c0 + 8 <= cs.C) {
      if (hi >= 0 && hi < cs.H && wi >= 0 && wi < cs.W) {
        const __bf16* src =
            X + (((long)n * cs.H + hi) * cs.W + wi) * cs.C + c0;
        vv = *(const bf16x8*)src;
      }
      return;
    }
    // generic path: the 8 taps may straddle (r,s) blocks (small C)
    int c = c0, rr = r, ss = s;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float v = 0.f;
      if (q0 + j < KD) {
        const int hij = ho * cs.U + rr - cs.P;
        const int wij = wo * cs.V + ss - cs.Q;
        if (hij >= 0 && hij < cs.H && wij >= 0 && wij < cs.W)
          v = (float)X[(((long)n * cs.H + hij) * cs.W + wij) * cs.C + c];
      }
      vv[j] = (__bf16)v;
      if (++c == cs.C) { c = 0; if (++ss == cs.S) { ss = 0; ++rr; } }
    }
  }

  DEVINL void commit(__bf16* Sm) {
    *(bf16x8*)sptr(Sm, mx, kk0) = vv;
  }
};

// Bs[k][kk]: weight memory [K][R*S*C] rows contiguous in tap order.
// Split load/commit like the patch stagers.
struct WrowStage {
  int x, kk0;
  bf16x8 vv;

  DEVINL void init(int t) {
    x = t >> 2;
    kk0 = (t & 3) * 8;
  }

  DEVINL void load(const __bf16* __restrict__ Wt, int n0, int k0,
                   int NROWS, int KD, bool vec) {
    const int gx = n0 + x;
    const int gk = k0 + kk0;
    const __bf16* src = Wt + (long)gx * KD + gk;
    vv = {};
    if (gx < NROWS && gk + 8 <= KD) {
      if (vec && ((gk & 7) == 0) && ((KD & 7) == 0)) {
        vv = *(const bf16x8*)src;
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) vv[j] = src[j];
      }
    } else if (gx < NROWS) {
#pragma unroll
      for (int j = 0; j < 8; ++j)
        if (gk + j < KD) vv[j] = src[j];
    }
  }

  DEVINL void commit(__bf16* Sn) {
    *(bf16x8*)sptr(Sn, x, kk0) = vv;
  }
};

// SK: split the tap reduction over grid.z (small-spatial deep layers
// otherwise leave most of the 256 CUs idle); slices store fp32 partial
// stripes, conv_reduce_kernel sums them in fixed order + applies the
// epilogue — same two-phase pattern as gemm.hip split-K.
template <bool BIAS, bool RELU, bool SK>
__global__ __launch_bounds__(256)
void conv_fwd_kernel(const __bf16* __restrict__ X, const __bf16* __restrict__ Wt,
                     const float* __restrict__ bias, __bf16* __restrict__ Y,
                     float* __restrict__ ws, int kc, ConvShape cs) {
  // double-buffered LDS: stage tile i+1 while MFMA consumes tile i —
  // one barrier per K-iteration, global-load latency overlapped
  __shared__ __align__(16) __bf16 As[2][TILE_ELEMS];
  __shared__ __align__(16) __bf16 Bs[2][TILE_ELEMS];
  const long M = (long)cs.N * cs.Ho * cs.Wo;
  const int KD = cs.R * cs.S * cs.C;
  const bool cvec = (cs.C & 7) == 0;
  const long tm0 = (long)blockIdx.x * BM;   // pixel tiles ride grid.x (2^31)
  const int tn0 = blockIdx.y * BN;
  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const int wr = wave >> 1, wc = wave & 1;

  const int ks = SK ? blockIdx.z * kc : 0;
  const int ke = SK ? min(ks + kc, KD) : KD;

  // 32x32x16 MFMA: one 32x32 wave tile, 2 MFMA + 4 fragment loads per
  // BK=32 iteration (vs 4 MFMA + 6 loads with 16x16x32 fragments).
  // A/B: lane l holds row/col (l&31), k = (l>>5)*8+j.
  // D: reg v -> row 8*(v>>2) + 4*(l>>5) + (v&3), col (l&31).
  f32x16 acc = {};
  FwdPatchStage pst;
  WrowStage wst;
  pst.init(cs, tm0, M, t);
  wst.init(t);
  pst.load(X, cs, ks, KD, cvec);
  wst.load(Wt, tn0, ks, cs.K, KD, true);
  pst.commit(As[0]);
  wst.commit(Bs[0]);
  __syncthreads();
  int cur = 0;
  for (int k0 = ks; k0 < ke; k0 += BK, cur ^= 1) {
    const bool more = k0 + BK < ke;
    if (more) {
      pst.load(X, cs, k0 + BK, KD, cvec);
      wst.load(Wt, tn0, k0 + BK, cs.K, KD, true);
    }
#pragma unroll
    for (int kh = 0; kh < 2; ++kh) {
      const int ko = kh * 16 + ((lane >> 5) << 3);
      bf16x8 a = *(const bf16x8*)sptr(As[cur], wr * 32 + (lane & 31), ko);
      bf16x8 b = *(const bf16x8*)sptr(Bs[cur], wc * 32 + (lane & 31), ko);
      acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc, 0, 0, 0);
    }
    if (more) {
      pst.commit(As[cur ^ 1]);
      wst.commit(Bs[cur ^ 1]);
    }
    __syncthreads();
  }

  const int k = tn0 + wc * 32 + (lane & 31);
  if (SK) {
    // store this slice's fp32 partial stripe
    if (k < cs.K) {
      float* wsl = ws + (long)blockIdx.z * M * cs.K;
#pragma unroll
      for (int v = 0; v < 16; ++v) {
        const long pm = tm0 + wr * 32 + ((v >> 2) << 3) +
                        ((lane >> 5) << 2) + (v & 3);
        if (pm < M) wsl[pm * cs.K + k] = acc[v];
      }
    }
    return;
  }
  // epilogue: Y[n,ho,wo,k] — k contiguous across lanes (coalesced)
  if (k < cs.K) {
    const float bv = BIAS ? bias[k] : 0.f;
#pragma unroll
    for (int v = 0; v < 16; ++v) {
      const long pm = tm0 + wr * 32 + ((v >> 2) << 3) +
                      ((lane >> 5) << 2) + (v & 3);
      if (pm >= M) continue;
      float val = acc[v] + bv;
      if (RELU) val = val > 0.f ? val : 0.f;
      Y[pm * cs.K + k] = (__bf16)val;
    }
  }
}

// split-KD phase 2: y[i] = act(sum_z ws[z][i] + bias[i % K])
template <bool BIAS, bool RELU, bool OUTF32 = false>
__global__ __launch_bounds__(256)
void conv_reduce_kernel(const float* __restrict__ ws,
                        const float* __restrict__ bias,
                        void* __restrict__ Yv, long mk, int K, int z) {
  __bf16* Y = (__bf16*)Yv;
  float* Yf = (float*)Yv;
  const long i0 = ((long)blockIdx.x * 256 + threadIdx.x) * 4;
  if (i0 >= mk) return;
  f32x4 v = {};
  for (int s = 0; s < z; ++s) {
    const float* p = ws + (long)s * mk + i0;
    if (i0 + 4 <= mk) {
      const f32x4 sv = *(const f32x4*)p;
#pragma unroll
      for (int j = 0; j < 4; ++j) v[j] += sv[j];
    } else {
      for (int j = 0; i0 + j < mk; ++j) v[j] += p[j];
    }
  }
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    const long i = i0 + j;
    if (i >= mk) break;
    float x = v[j];
    if (BIAS) x += bias[i % K];
    if (RELU) x = x > 0.f ? x : 0.f;
    if (OUTF32) Yf[i] = x;
    else Y[i] = (__bf16)x;
  }
}

// --------------------------------------------------------------- bwd-data

// As[m][kk]: m = input pixel, kk = tap (r,s,k) with k innermost, reading
// dY[n, (h+P-r)/U, (w+Q-s)/V, k] (contiguous in k). STRIDE specializes
// the common cases: 1 removes the divisibility tests, 2 turns them into
// bit ops (runtime %/ by the stride costs ~30 VALU cycles each).
// Pixel decode (n,h,w) hoisted per thread (the m-tile is fixed for the
// whole tap loop) — 2 fdivs per tile instead of 4.
struct BwddPatchStage {
  int mx, kk0;
  int n, h, w;
  bool pok;

  DEVINL void init(const ConvShape& cs, long m0, long M, int t) {
    mx = t >> 2;
    kk0 = (t & 3) * 8;
    const long pm = m0 + mx;
    pok = pm < M;
    const unsigned pc = (unsigned)(pok ? pm : 0);
    const int HWi = cs.H * cs.W;
    n = (int)fdiv(pc, cs.dHW);
    const int rem = (int)(pc - (unsigned)n * HWi);
    h = fdiv(rem, cs.dW2);
    w = rem - h * cs.W;
  }

  bf16x8 vv;   // in-flight tile (load() -> commit())

  template <int STRIDE>
  DEVINL void load(const __bf16* __restrict__ dY, const ConvShape cs,
                   int k0, int KD, bool kvec);

  DEVINL void commit(__bf16* Sm) {
    *(bf16x8*)sptr(Sm, mx, kk0) = vv;
  }
};

// split load/commit (see FwdPatchStage): global reads land in
// registers before the MFMA block; the b128 LDS store runs after it.
template <int STRIDE>
DEVINL void BwddPatchStage::load(const __bf16* __restrict__ dY,
                                 const ConvShape cs,
                                 int k0, int KD, bool kvec) {
  const int q0 = k0 + kk0;
  vv = {};
  if (q0 >= KD || !pok) return;
  const int rs = fdiv(q0, cs.dK);
  const int kc0 = q0 - rs * cs.K;
  const int r = fdiv(rs, cs.dS);
  const int s = rs - r * cs.S;
  const int hn = h + cs.P - r, wn = w + cs.Q - s;

  if (kvec && kc0 + 8 <= cs.K) {
    bool ok;
    int ho, wo;
    if (STRIDE == 1) {
      ho = hn; wo = wn;
      ok = hn >= 0 && hn < cs.Ho && wn >= 0 && wn < cs.Wo;
    } else if (STRIDE == 2) {
      ok = hn >= 0 && wn >= 0 && !(hn & 1) && !(wn & 1);
      ho = hn >> 1; wo = wn >> 1;
      ok = ok && ho < cs.Ho && wo < cs.Wo;
    } else {
      ok = hn >= 0 && wn >= 0 && hn % cs.U == 0 && wn % cs.V == 0;
      ho = hn / cs.U; wo = wn / cs.V;
      ok = ok && ho < cs.Ho && wo < cs.Wo;
    }
    if (ok) {
      const __bf16* src =
          dY + (((long)n * cs.Ho + ho) * cs.Wo + wo) * cs.LDY + kc0;
      vv = *(const bf16x8*)src;
    }
    return;
  }
  int kc = kc0, rr2 = r, ss2 = s;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    float v = 0.f;
    if (q0 + j < KD) {
      const int hnj = h + cs.P - rr2, wnj = w + cs.Q - ss2;
      bool ok;
      int ho, wo;
      if (STRIDE == 1) {
        ho = hnj; wo = wnj;
        ok = hnj >= 0 && hnj < cs.Ho && wnj >= 0 && wnj < cs.Wo;
      } else if (STRIDE == 2) {
        ok = hnj >= 0 && wnj >= 0 && !(hnj & 1) && !(wnj & 1);
        ho = hnj >> 1; wo = wnj >> 1;
        ok = ok && ho < cs.Ho && wo < cs.Wo;
      } else {
        ok = hnj >= 0 && wnj >= 0 && hnj % cs.U == 0 && wnj % cs.V == 0;
        ho = hnj / cs.U; wo = wnj / cs.V;
        ok = ok && ho < cs.Ho && wo < cs.Wo;
      }
      if (ok)
        v = (float)dY[(((long)n * cs.Ho + ho) * cs.Wo + wo) * cs.LDY + kc];
    }
    vv[j] = (__bf16)v;
    if (++kc == cs.K) { kc = 0; if (++ss2 == cs.S) { ss2 = 0; ++rr2; } }
  }
}

// Bs[c][q]: weight read in its NATIVE [K,R,S,C] layout (the forward
// tensor) — 8 contiguous c for one tap q=(r,s,k), transposed into LDS.
// Kills the host-side W^T permute+copy the backward used to pay per
// layer per step (~94 launch-bound copies across Inception).
// Split load/commit like the patch stagers.
// gfx950 ds_read_b64_tr_b16 semantics (measured, tools/trprobe.hip —
// all four probe patterns fit): within each aligned 16-lane group,
// provider lane x reads 8 B (4 bf16) at its own 8 B-aligned address
// forming a 16x4 element matrix M, and consumer lane l receives
// elem j = M[4*j + ((l>>2)&3)][l&3]. Feeding MFMA lane l with
// reduction rows ko..ko+3 of its column m = 32*wr + (l&31) therefore
// takes provider addresses  addr(l) = (ko + ((l&15)>>2))*TR_L +
// 32*wr + 16*((l>>4)&1) + 4*(l&3)  — four consecutive image ROWS per
// group, and a second read at +4*TR_L covers rows ko+4..ko+7.
//
// Row stride TR_L = 96 elems (192 B ≡ 64 banks*3 mod 256 B): the four
// rows a 32-lane service group touches land on bank quartets 0-15 /
// 48-63 / 32-47 / 16-31 — a partition, so both tr reads and the b128
// staging writes are conflict-free.
constexpr int TR_L = 96;
constexpr int TR_ELEMS = BK * TR_L;

DEVINL int trindex(int p, int c) { return p * TR_L + c; }

DEVINL bf16x4 tr_read(const __bf16* p) {
  return __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
      (__attribute__((address_space(3))) bf16x4*)p);
}

struct WkrscStage {
  int px, kk0;
  bf16x8 vv;

  DEVINL void init(int t) {
    px = t >> 3;          // tap q within the BK tile (0..31)
    kk0 = (t & 7) * 8;    // c chunk
  }

  DEVINL void load(const __bf16* __restrict__ W, int c0blk, int q0,
                   const ConvShape cs, int KD) {
    const int q = q0 + px;
    const int c = c0blk + kk0;
    vv = {};
    if (q >= KD || c >= cs.C) return;
    const int rs = fdiv(q, cs.dK);
    const int k = q - rs * cs.K;
    const int r = fdiv(rs, cs.dS);
    const int s = rs - r * cs.S;
    const __bf16* src = W + (((long)k * cs.R + r) * cs.S + s) * cs.C + c;
    if (((cs.C & 7) == 0) && c + 8 <= cs.C) {
      vv = *(const bf16x8*)src;
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j)
        if (c + j < cs.C) vv[j] = src[j];
    }
  }

  DEVINL void commit(__bf16* Sn) {
#pragma unroll
    for (int j = 0; j < 8; ++j) *sptr(Sn, kk0 + j, px) = vv[j];
  }
};

// Reduction-major variant of WkrscStage: one b128 LDS store into the
// [tap][c] image (WkrscStage pays 8 sub-dword scatter writes to
// transpose); the MFMA B fragments come back via ds_read_b64_tr_b16.
struct WkrscTrStage : WkrscStage {
  int so;
  DEVINL void init(int t) {
    WkrscStage::init(t);
    so = trindex(px, kk0);
  }
  DEVINL void commit(__bf16* Sn) { *(bf16x8*)&Sn[so] = vv; }
};

template <int STRIDE, bool SK, bool TRB>
__global__ __launch_bounds__(256)
void conv_bwdd_kernel(const __bf16* __restrict__ dY, const __bf16* __restrict__ Wt,
                      __bf16* __restrict__ dX, float* __restrict__ ws, int kc,
                      ConvShape cs) {
  // Wt memory: NATIVE [K][R*S*C] (the forward weight tensor)
  __shared__ __align__(16) __bf16 As[2][TILE_ELEMS];
  __shared__ __align__(16) __bf16 Bs[2][TRB ? TR_ELEMS : TILE_ELEMS];
  const long M = (long)cs.N * cs.H * cs.W;
  const int KD = cs.R * cs.S * cs.K;
  const bool kvec = (cs.K & 7) == 0;
  const long tm0 = (long)blockIdx.x * BM;
  const int tn0 = blockIdx.y * BN;
  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const int wr = wave >> 1, wc = wave & 1;

  const int ks = SK ? blockIdx.z * kc : 0;
  const int ke = SK ? min(ks + kc, KD) : KD;

  // 32x32x16 MFMA core (same scheme as conv_fwd: one 32x32 wave tile,
  // 2 MFMA + 2 fragment loads per BK=32 iteration — ~1.7x the issue
  // efficiency of the earlier 2x2 16x16x32 fragment scheme here).
  f32x16 acc = {};
  BwddPatchStage pst;
  typename std::conditional<TRB, WkrscTrStage, WkrscStage>::type wst;
  pst.init(cs, tm0, M, t);
  wst.init(t);
  // provider-lane fragment address for the tr image (unused when !TRB)
  const int bbase = (((lane & 15) >> 2) + (lane >> 5) * 8) * TR_L +
                    wc * 32 + 16 * ((lane >> 4) & 1) + 4 * (lane & 3);
  pst.load<STRIDE>(dY, cs, ks, KD, kvec);
  wst.load(Wt, tn0, ks, cs, KD);
  pst.commit(As[0]);
  wst.commit(Bs[0]);
  __syncthreads();
  int cur = 0;
  for (int k0 = ks; k0 < ke; k0 += BK, cur ^= 1) {
    const bool more = k0 + BK < ke;
    if (more) {
      pst.load<STRIDE>(dY, cs, k0 + BK, KD, kvec);
      wst.load(Wt, tn0, k0 + BK, cs, KD);
    }
#pragma unroll
    for (int kh = 0; kh < 2; ++kh) {
      const int ko = kh * 16 + ((lane >> 5) << 3);
      bf16x8 a = *(const bf16x8*)sptr(As[cur], wr * 32 + (lane & 31), ko);
      bf16x8 b;
      if (TRB) {
        bf16x4 b0 = tr_read(&Bs[cur][bbase + kh * 16 * TR_L]);
        bf16x4 b1 = tr_read(&Bs[cur][bbase + kh * 16 * TR_L + 4 * TR_L]);
        b = __builtin_shufflevector(b0, b1, 0, 1, 2, 3, 4, 5, 6, 7);
      } else {
        b = *(const bf16x8*)sptr(Bs[cur], wc * 32 + (lane & 31), ko);
      }
      acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc, 0, 0, 0);
    }
    if (more) {
      pst.commit(As[cur ^ 1]);
      wst.commit(Bs[cur ^ 1]);
    }
    __syncthreads();
  }

  const int c = tn0 + wc * 32 + (lane & 31);
  if (SK) {
    if (c < cs.C) {
      float* wsl = ws + (long)blockIdx.z * M * cs.C;
#pragma unroll
      for (int v = 0; v < 16; ++v) {
        const long pm = tm0 + wr * 32 + ((v >> 2) << 3) +
                        ((lane >> 5) << 2) + (v & 3);
        if (pm < M) wsl[pm * cs.C + c] = acc[v];
      }
    }
    return;
  }
  if (c < cs.C) {
#pragma unroll
    for (int v = 0; v < 16; ++v) {
      const long pm = tm0 + wr * 32 + ((v >> 2) << 3) +
                      ((lane >> 5) << 2) + (v & 3);
      if (pm >= M) continue;
      dX[pm * cs.C + c] = (__bf16)acc[v];
    }
  }
}

// -------------------------------------------------------------- bwd-weight

// As[k][p]: dY[n,ho,wo,k] — thread t loads 8 consecutive k (one b128)
// for one reduction pixel p, transposing into LDS.
// COLUMN-wise transpose-free staging: thread t owns ONE k row
// (t & 63) and an 8-pixel chunk ((t >> 6) * 8); the 8 global loads are
// lane-coalesced across k (consecutive addresses within an
// instruction) and the LDS store is a single aligned b128 into row k —
// the same consecutive-row/same-column pattern as the MFMA fragment
// reads, which the 8-row skew already makes conflict-free. The earlier
// row-wise variant (one b128 load, 8 transposed b16 stores) serialized
// on sub-dword LDS write conflicts (PMC: conflict count exceeded the
// LDS instruction count in this kernel).
struct DyBwdwStage {
  int k, pxc;
  bool kok;
  long p;              // first pixel of this thread's chunk
  const __bf16* src;   // advances by BK*K per tile
  bf16x8 v;            // in-flight tile (load() -> commit())

  DEVINL void init(const __bf16* __restrict__ dY, const ConvShape& cs,
                   int m0, long p0, int t, bool kvec) {
    (void)kvec;
    k = t & 63;            // LDS row (k within the tile)
    pxc = (t >> 6) * 8;    // 8-pixel chunk
    kok = m0 + k < cs.K;
    p = p0 + pxc;
    src = dY + p * cs.LDY + (m0 + k);
  }

  // split load/commit: load() issues the global reads into registers
  // BEFORE the MFMA block so their latency is covered by compute;
  // commit() (after the MFMAs) only pays the vmcnt drain + LDS write.
  // The fused form parked every wave on vmcnt ahead of its own MFMAs
  // (profiles/r01_inception_pmc.txt: SQ_WAIT 150x the MFMA count).
  // The 8 strided loads use j*K offsets off ONE base (a sequential
  // pointer bump made each load's ADDRESS depend on the previous —
  // PMC showed 42 VALU per MFMA, much of it serial address math).
  DEVINL void load(const ConvShape cs, long Ptot) {
    v = {};
    if (kok) {
      const long left = Ptot - p;
      if (left >= 8) {
#pragma unroll
        for (int j = 0; j < 8; ++j) v[j] = src[(long)j * cs.K];
      } else if (left > 0) {
        for (int j = 0; j < (int)left; ++j) v[j] = src[(long)j * cs.K];
      }
    }
    p += BK;
    src += (long)BK * cs.K;
  }

  DEVINL void commit(__bf16* Sm) {
    *(bf16x8*)sptr(Sm, k, pxc) = v;
  }
};

// Bs[tap(r,s,c)][p]: X[n, ho*U+r-P, wo*V+s-Q, c] — thread t loads 8
// consecutive c (one b128) for one pixel, transposing into LDS.
// The tap decode (r,s,c) is FIXED per thread for the whole reduction
// loop (only the pixel advances), so it is hoisted into init() and the
// pixel (n,rem) walks incrementally — one fdiv per tile instead of
// four (the PMC profile showed 21 VALU instructions per MFMA here).
struct XBwdwStage {
  int px, qq0, q0;
  int c0, r, s;      // hoisted tap decode
  bool tapok, vec8;
  long p;            // current reduction pixel = p0 + px
  int n;             // p = n*HoWo + rem
  int rem, HoWo;
  bf16x8 vv;         // in-flight tile (load() -> commit())

  DEVINL void init(const ConvShape& cs, int n0, long p0, int KD, int t,
                   bool cvec) {
    px = t >> 3;
    qq0 = (t & 7) * 8;
    q0 = n0 + qq0;
    tapok = q0 < KD;
    const int qc = tapok ? q0 : 0;
    const int rs = fdiv(qc, cs.dC);
    c0 = qc - rs * cs.C;
    r = fdiv(rs, cs.dS);
    s = rs - r * cs.S;
    vec8 = tapok && cvec && c0 + 8 <= cs.C;
    HoWo = cs.Ho * cs.Wo;
    p = p0 + px;
    n = (int)fdiv((unsigned)min(p, (long)0x7fffffff), cs.dHoWo);
    rem = (int)(p - (long)n * HoWo);
  }

  // split load/commit (see DyBwdwStage): global reads land in
  // registers before the MFMA block; the transposed LDS scatter waits
  // for them only AFTER the MFMAs are issued.
  DEVINL void load(const __bf16* __restrict__ X, const ConvShape cs,
                   int KD, long Ptot) {
    const int ho = fdiv(rem, cs.dWo);
    const int wo = rem - ho * cs.Wo;
    const bool pok = p < Ptot;
    vv = {};
    if (vec8) {
      const int hi = ho * cs.U + r - cs.P;
      const int wi = wo * cs.V + s - cs.Q;
      if (pok && hi >= 0 && hi < cs.H && wi >= 0 && wi < cs.W) {
        const __bf16* src =
            X + (((long)n * cs.H + hi) * cs.W + wi) * cs.C + c0;
        vv = *(const bf16x8*)src;
      }
    } else if (tapok && pok) {
      // generic path: the 8 taps may straddle (r,s) blocks (small C)
      int c = c0, rr2 = r, ss2 = s;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float v = 0.f;
        if (q0 + j < KD) {
          const int hij = ho * cs.U + rr2 - cs.P;
          const int wij = wo * cs.V + ss2 - cs.Q;
          if (hij >= 0 && hij < cs.H && wij >= 0 && wij < cs.W)
            v = (float)X[(((long)n * cs.H + hij) * cs.W + wij) * cs.C + c];
        }
        vv[j] = (__bf16)v;
        if (++c == cs.C) { c = 0; if (++ss2 == cs.S) { ss2 = 0; ++rr2; } }
      }
    }
    p += BK;
    rem += BK;
    while (rem >= HoWo) { rem -= HoWo; ++n; }
  }

  DEVINL void commit(__bf16* Sn) {
    // transposed scatter: 8 rows, same column px (8-row skew keeps the
    // groups conflict-free)
#pragma unroll
    for (int j = 0; j < 8; ++j) *sptr(Sn, qq0 + j, px) = vv[j];
  }
};

// dW fp32 out [K][R*S*C]; grid.z slices the huge N*Ho*Wo reduction and
// accumulates with fp32 atomics (dW zero-filled by the binding when
// sliced). A fixed-order stripe-reduce variant measured SLOWER: dW is
// small (9-300K elements), so the z-deep serial reduce loop had too
// little parallelism, while the atomic contention here is negligible.
__global__ __launch_bounds__(256)
void conv_bwdw_kernel(const __bf16* __restrict__ dY, const __bf16* __restrict__ X,
                      float* __restrict__ dW, float* __restrict__ ws,
                      ConvShape cs, long pc) {
  __shared__ __align__(16) __bf16 As[2][TILE_ELEMS];
  __shared__ __align__(16) __bf16 Bs[2][TILE_ELEMS];
  const int KD = cs.R * cs.S * cs.C;
  const long Ptot = (long)cs.N * cs.Ho * cs.Wo;
  const bool cvec = (cs.C & 7) == 0;
  const bool kvec = (cs.K & 7) == 0;
  const int tm0 = blockIdx.y * BM;
  const int tn0 = blockIdx.x * BN;
  const long ps = (long)blockIdx.z * pc;
  const long pe = min(ps + pc, Ptot);
  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const int wr = wave >> 1, wc = wave & 1;

  // 32x32x16 MFMA core (same scheme as conv_fwd/bwdd: one 32x32 wave
  // tile, 2 MFMA + 4 fragment reads per BK=32 iteration — vs 4 MFMA +
  // 6 reads with the earlier 2x2 16x16x32 fragments).
  f32x16 acc = {};
  DyBwdwStage dst;
  XBwdwStage xst;
  dst.init(dY, cs, tm0, ps, t, kvec);
  xst.init(cs, tn0, ps, KD, t, cvec);
  dst.load(cs, Ptot);
  xst.load(X, cs, KD, Ptot);
  dst.commit(As[0]);
  xst.commit(Bs[0]);
  __syncthreads();
  int cur = 0;
  for (long p0 = ps; p0 < pe; p0 += BK, cur ^= 1) {
    const bool more = p0 + BK < pe;
    if (more) {
      // next tile's global reads first: their latency is covered by
      // this tile's fragment reads + MFMAs (the fused stage parked on
      // vmcnt before any MFMA issued)
      dst.load(cs, Ptot);
      xst.load(X, cs, KD, Ptot);
    }
#pragma unroll
    for (int kh = 0; kh < 2; ++kh) {
      const int ko = kh * 16 + ((lane >> 5) << 3);
      bf16x8 a = *(const bf16x8*)sptr(As[cur], wr * 32 + (lane & 31), ko);
      bf16x8 b = *(const bf16x8*)sptr(Bs[cur], wc * 32 + (lane & 31), ko);
      acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc, 0, 0, 0);
    }
    if (more) {
      dst.commit(As[cur ^ 1]);
      xst.commit(Bs[cur ^ 1]);
    }
    __syncthreads();
  }

  // D: reg v -> A-row (k) 8*(v>>2) + 4*(lane>>5) + (v&3), B-col (q)
  // lane&31 — q contiguous across lanes, so the dW stores coalesce
  const int q = tn0 + wc * 32 + (lane & 31);
  if (q < KD) {
#pragma unroll
    for (int v = 0; v < 16; ++v) {
      const int k = tm0 + wr * 32 + ((v >> 2) << 3) +
                    ((lane >> 5) << 2) + (v & 3);
      if (k >= cs.K) continue;
      if (gridDim.z == 1)
        dW[(long)k * KD + q] = acc[v];
      else
        unsafeAtomicAdd(&dW[(long)k * KD + q], acc[v]);
    }
  }
}

// --------------------------------------------- bwd-weight, transpose-read
// Pixel-major LDS image + ds_read_b64_tr_b16 fragment reads.
//
// The row-major (k-major) image above forces the staging to transpose
// somewhere: DyBwdwStage pays 8 strided b16 GLOBAL loads per thread per
// tile (k-coalesced but 8 separate instructions), XBwdwStage pays 8
// sub-dword LDS scatter writes. Pixel-major storage makes BOTH stagers
// a single b128 global load + a single b128 LDS write per thread per
// tile, and the MFMA A/B fragments (8 reduction pixels for one column)
// come out of the plain [pixel][column] image via two transpose-reads
// each (tr reads cost no extra issue slots beside MFMAs —
// MI355X_MICROARCH.md filler table).
//
// gfx950 ds_read_b64_tr_b16 semantics (measured, tools/trprobe.hip —
// all four probe patterns fit): within each aligned 16-lane group,
// provider lane x reads 8 B (4 bf16) at its own 8 B-aligned address
// forming a 16x4 element matrix M, and consumer lane l receives
// elem j = M[4*j + ((l>>2)&3)][l&3]. Feeding MFMA lane l with pixels
// ko..ko+3 of its column m = 32*wr + (l&31) therefore takes provider
// addresses  addr(l) = (ko + ((l&15)>>2))*TR_L + 32*wr +
// 16*((l>>4)&1) + 4*(l&3)  — four consecutive image ROWS per group,
// and a second read at +4*TR_L covers pixels ko+4..ko+7.
//
// Row stride TR_L = 96 elems (192 B ≡ 64 banks*3 mod 256 B): the four
// rows a 32-lane service group touches land on bank quartets 0-15 /
// 48-63 / 32-47 / 16-31 — a partition, so both tr reads and the b128
// staging writes are conflict-free.
// As[p][k]: dY[n,ho,wo,k] — thread t owns ONE pixel (t>>3) and an
// 8-channel chunk ((t&7)*8): the global load is one b128 (8 consecutive
// k of one dY row) and the LDS store is one b128 into the pixel-major
// image. Compare DyBwdwStage: same bytes, 8 strided loads + transposed
// store.
struct DyBwdwTrStage {
  int so;              // LDS elem offset (fixed per thread)
  int k0;
  bool vec;
  long p;
  const __bf16* src;
  bf16x8 v;

  DEVINL void init(const __bf16* __restrict__ dY, const ConvShape& cs,
                   int m0, long p0, int t, bool kvec) {
    const int px = t >> 3, kc0 = (t & 7) * 8;
    so = trindex(px, kc0);
    k0 = m0 + kc0;
    vec = kvec && (k0 + 8 <= cs.K);
    p = p0 + px;
    src = dY + p * cs.LDY + k0;
  }

  DEVINL void load(const ConvShape cs, long Ptot) {
    v = {};
    if (p < Ptot) {
      if (vec) {
        v = *(const bf16x8*)src;
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j)
          if (k0 + j < cs.K) v[j] = src[j];
      }
    }
    p += BK;
    src += (long)BK * cs.LDY;
  }

  DEVINL void commit(__bf16* Sm) { *(bf16x8*)&Sm[so] = v; }
};

// Bs[p][q]: X patches — identical thread mapping and load path to
// XBwdwStage (one pixel, 8 consecutive taps), but the store is one
// b128 into the pixel-major image instead of 8 b16 scatters.
struct XBwdwTrStage : XBwdwStage {
  int so;
  DEVINL void init_tr(const ConvShape& cs, int n0, long p0, int KD, int t,
                      bool cvec) {
    init(cs, n0, p0, KD, t, cvec);
    so = trindex(px, qq0);
  }
  DEVINL void commit_tr(__bf16* Sn) { *(bf16x8*)&Sn[so] = vv; }
};

template <int MINB>
__global__ __launch_bounds__(256, MINB)
void conv_bwdw_kernel_tr(const __bf16* __restrict__ dY,
                         const __bf16* __restrict__ X,
                         float* __restrict__ dW, ConvShape cs, long pc) {
  __shared__ __align__(16) __bf16 As[2][TR_ELEMS];
  __shared__ __align__(16) __bf16 Bs[2][TR_ELEMS];
  const int KD = cs.R * cs.S * cs.C;
  const long Ptot = (long)cs.N * cs.Ho * cs.Wo;
  const bool cvec = (cs.C & 7) == 0;
  // b128 dY loads need the 8-k chunk 16 B aligned: base + p*LDY + k0
  const bool kvec = ((cs.K & 7) == 0) && ((cs.LDY & 7) == 0) &&
                    ((((unsigned long long)dY) & 15ULL) == 0);
  const int tm0 = blockIdx.y * BM;
  const int tn0 = blockIdx.x * BN;
  const long ps = (long)blockIdx.z * pc;
  const long pe = min(ps + pc, Ptot);
  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const int wr = wave >> 1, wc = wave & 1;

  f32x16 acc = {};
  DyBwdwTrStage dst;
  XBwdwTrStage xst;
  dst.init(dY, cs, tm0, ps, t, kvec);
  xst.init_tr(cs, tn0, ps, KD, t, cvec);
  dst.load(cs, Ptot);
  xst.load(X, cs, KD, Ptot);
  dst.commit(As[0]);
  xst.commit_tr(Bs[0]);
  __syncthreads();

  // provider-lane fragment addresses (fixed per lane; see header
  // comment): row = pixel within the tile, column = this lane group's
  // 4-element slice of the output column block
  const int arow0 = ((lane & 15) >> 2) + (lane >> 5) * 8;
  const int acoff = 16 * ((lane >> 4) & 1) + 4 * (lane & 3);
  const int abase = arow0 * TR_L + wr * 32 + acoff;
  const int bbase = arow0 * TR_L + wc * 32 + acoff;

  int cur = 0;
  for (long p0 = ps; p0 < pe; p0 += BK, cur ^= 1) {
    const bool more = p0 + BK < pe;
    if (more) {
      dst.load(cs, Ptot);
      xst.load(X, cs, KD, Ptot);
    }
#pragma unroll
    for (int kh = 0; kh < 2; ++kh) {
      const int tp = kh * 16 * TR_L;
      bf16x4 a0 = tr_read(&As[cur][abase + tp]);
      bf16x4 a1 = tr_read(&As[cur][abase + tp + 4 * TR_L]);
      bf16x4 b0 = tr_read(&Bs[cur][bbase + tp]);
      bf16x4 b1 = tr_read(&Bs[cur][bbase + tp + 4 * TR_L]);
      bf16x8 a = __builtin_shufflevector(a0, a1, 0, 1, 2, 3, 4, 5, 6, 7);
      bf16x8 b = __builtin_shufflevector(b0, b1, 0, 1, 2, 3, 4, 5, 6, 7);
      acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc, 0, 0, 0);
    }
    if (more) {
      dst.commit(As[cur ^ 1]);
      xst.commit_tr(Bs[cur ^ 1]);
    }
    __syncthreads();
  }

  const int q = tn0 + wc * 32 + (lane & 31);
  if (q < KD) {
#pragma unroll
    for (int v = 0; v < 16; ++v) {
      const int k = tm0 + wr * 32 + ((v >> 2) << 3) +
                    ((lane >> 5) << 2) + (v & 3);
      if (k >= cs.K) continue;
      if (gridDim.z == 1)
        dW[(long)k * KD + q] = acc[v];
      else
        unsafeAtomicAdd(&dW[(long)k * KD + q], acc[v]);
    }
  }
}

}  // namespace

int conv_fwd_slices(int N, int K, int Ho, int Wo, int C, int R, int S) {
  // binding uses this to size the split workspace; must match launch
  const long M = (long)N * Ho * Wo;
  const long tiles = ((M + BM - 1) / BM) * ceil_div(K, BN);
  const int KD = R * S * C;
  if (tiles >= 512 || KD < 512) return 1;
  long zwant = (1024 + tiles - 1) / tiles;
  long zmax = KD / (2 * BK);
  long z = zwant < zmax ? zwant : zmax;
  if (z > 16) z = 16;
  if (z < 1) z = 1;
  return (int)z;
}

int conv_bwdd_slices(int N, int H, int W, int C, int K, int R, int S) {
  const long M = (long)N * H * W;
  const long tiles = ((M + BM - 1) / BM) * ceil_div(C, BN);
  const int KD = R * S * K;
  if (tiles >= 512 || KD < 512) return 1;
  long zwant = (1024 + tiles - 1) / tiles;
  long zmax = KD / (2 * BK);
  long z = zwant < zmax ? zwant : zmax;
  if (z > 16) z = 16;
  if (z < 1) z = 1;
  return (int)z;
}

void launch_conv_fwd(const bf16_t* X, const bf16_t* W, const float* bias,
                     bf16_t* Y, float* ws, int N, int C, int H, int Wd,
                     int K, int R, int S, int Ho, int Wo, int U, int V,
                     int P, int Q, bool relu, hipStream_t stream) {
  ConvShape cs{N, C, H, Wd, K, R, S, Ho, Wo, U, V, P, Q,
               make_fdiv(C), make_fdiv(S), make_fdiv(K), make_fdiv(Wo),
               make_fdiv(Wd), make_fdiv(Ho * Wo), make_fdiv(H * Wd), K};
  const long M = (long)N * Ho * Wo;
  const int KD = R * S * C;
  const int z = conv_fwd_slices(N, K, Ho, Wo, C, R, S);
  dim3 block(256);
  if (z > 1 && ws) {
    int kc = ceil_div(ceil_div(KD, z), BK) * BK;
    const int zr = ceil_div(KD, kc);
    dim3 grid((unsigned)((M + BM - 1) / BM), ceil_div(K, BN), zr);
    hipLaunchKernelGGL((conv_fwd_kernel<false, false, true>), grid, block,
                       0, stream, (const __bf16*)X, (const __bf16*)W, bias,
                       (__bf16*)Y, ws, kc, cs);
    const long mk = M * K;
    dim3 rgrid((unsigned)((mk / 4 + 255) / 256)), rblock(256);
#define RL(BIASv, RELUv)                                                    \
    hipLaunchKernelGGL((conv_reduce_kernel<BIASv, RELUv>), rgrid, rblock,   \
                       0, stream, ws, bias, (__bf16*)Y, mk, K, zr)
    if (bias) { if (relu) RL(true, true); else RL(true, false); }
    else      { if (relu) RL(false, true); else RL(false, false); }
#undef RL
    return;
  }
  dim3 grid((unsigned)((M + BM - 1) / BM), ceil_div(K, BN));
#define FL(BIASv, RELUv)                                                    \
  hipLaunchKernelGGL((conv_fwd_kernel<BIASv, RELUv, false>), grid, block,   \
                     0, stream, (const __bf16*)X, (const __bf16*)W, bias,   \
                     (__bf16*)Y, nullptr, 0, cs)
  if (bias) { if (relu) FL(true, true); else FL(true, false); }
  else      { if (relu) FL(false, true); else FL(false, false); }
#undef FL
}

static int bwdd_tr() {
  static int m = -1;
  if (m < 0) {
    const char* e = getenv("TFA_BWDD_TR");
    m = e ? atoi(e) : 1;
  }
  return m;
}

void launch_conv_bwd_data(const bf16_t* dY, long ldy, const bf16_t* Wt,
                          bf16_t* dX,
                          float* ws, int N, int C, int H, int Wd, int K,
                          int R, int S, int Ho, int Wo, int U, int V, int P,
                          int Q, hipStream_t stream) {
  ConvShape cs{N, C, H, Wd, K, R, S, Ho, Wo, U, V, P, Q,
               make_fdiv(C), make_fdiv(S), make_fdiv(K), make_fdiv(Wo),
               make_fdiv(Wd), make_fdiv(Ho * Wo), make_fdiv(H * Wd),
               (int)ldy};
  const long M = (long)N * H * Wd;
  const int KD = R * S * K;
  const int z = conv_bwdd_slices(N, H, Wd, C, K, R, S);
  dim3 block(256);
  if (z > 1 && ws) {
    int kc = ceil_div(ceil_div(KD, z), BK) * BK;
    const int zr = ceil_div(KD, kc);
    dim3 grid((unsigned)((M + BM - 1) / BM), ceil_div(C, BN), zr);
#define BL(STRIDEv)                                                         \
    do { if (bwdd_tr())                                                     \
      hipLaunchKernelGGL((conv_bwdd_kernel<STRIDEv, true, true>), grid,     \
                         block, 0, stream, (const __bf16*)dY,               \
                         (const __bf16*)Wt, (__bf16*)dX, ws, kc, cs);       \
    else                                                                    \
      hipLaunchKernelGGL((conv_bwdd_kernel<STRIDEv, true, false>), grid,    \
                         block, 0, stream, (const __bf16*)dY,               \
                         (const __bf16*)Wt, (__bf16*)dX, ws, kc, cs);       \
    } while (0)
    if (U == 1 && V == 1) BL(1);
    else if (U == 2 && V == 2) BL(2);
    else BL(0);
#undef BL
    const long mk = M * C;
    dim3 rgrid((unsigned)((mk / 4 + 255) / 256)), rblock(256);
    hipLaunchKernelGGL((conv_reduce_kernel<false, false>), rgrid, rblock, 0,
                       stream, ws, nullptr, (__bf16*)dX, mk, C, zr);
    return;
  }
  dim3 grid((unsigned)((M + BM - 1) / BM), ceil_div(C, BN));
#define BL(STRIDEv)                                                         \
  do { if (bwdd_tr())                                                       \
    hipLaunchKernelGGL((conv_bwdd_kernel<STRIDEv, false, true>), grid,      \
                       block, 0, stream, (const __bf16*)dY,                 \
                       (const __bf16*)Wt, (__bf16*)dX, nullptr, 0, cs);     \
  else                                                                      \
    hipLaunchKernelGGL((conv_bwdd_kernel<STRIDEv, false, false>), grid,     \
                       block, 0, stream, (const __bf16*)dY,                 \
                       (const __bf16*)Wt, (__bf16*)dX, nullptr, 0, cs);     \
  } while (0)
  if (U == 1 && V == 1) BL(1);
  else if (U == 2 && V == 2) BL(2);
  else BL(0);
#undef BL
}

int conv_bwdw_slices(int N, int C, int K, int R, int S, int Ho, int Wo) {
  const int KD = C * R * S;
  const long Ptot = (long)N * Ho * Wo;
  const long tiles = (long)ceil_div(K, BM) * ceil_div(KD, BN);
  long zmax = (Ptot + BK - 1) / BK;
  long zwant = 1024 / tiles;
  if (zwant < 1) zwant = 1;
  if (zwant > 256) zwant = 256;   // ws stays bounded: z*tiles ~ 1024
  int z = (int)(zmax < zwant ? zmax : zwant);
  long pc = (Ptot + z - 1) / z;
  pc = (pc + BK - 1) / BK * BK;
  return (int)((Ptot + pc - 1) / pc);
}

void launch_conv_bwd_weight(const bf16_t* dY, long ldy, const bf16_t* X,
                            float* dW,
                            int N, int C, int H, int Wd, int K,
                            int R, int S, int Ho, int Wo, int U, int V,
                            int P, int Q, hipStream_t stream) {
  float* ws = nullptr;  // unused (atomic accumulate)
  (void)ws;
  ConvShape cs{N, C, H, Wd, K, R, S, Ho, Wo, U, V, P, Q,
               make_fdiv(C), make_fdiv(S), make_fdiv(K), make_fdiv(Wo),
               make_fdiv(Wd), make_fdiv(Ho * Wo), make_fdiv(H * Wd),
               (int)ldy};
  const int KD = C * R * S;
  const long Ptot = (long)N * Ho * Wo;
  int z = conv_bwdw_slices(N, C, K, R, S, Ho, Wo);
  long pc = (Ptot + z - 1) / z;
  pc = (pc + BK - 1) / BK * BK;
  z = (int)((Ptot + pc - 1) / pc);
  dim3 grid(ceil_div(KD, BN), ceil_div(K, BM), z);
  dim3 block(256);
  // TFA_BWDW_TR=0 falls back to the k-major staging kernel (A/B
  // comparison; the transpose-read variant measured faster across the
  // Inception shapes — docs/KERNELS.md round-2 log)
  static int tr_mode = -1;
  if (tr_mode < 0) {
    const char* e = getenv("TFA_BWDW_TR");
    tr_mode = e ? atoi(e) : 1;
  }
  if (tr_mode == 2) {
    hipLaunchKernelGGL((conv_bwdw_kernel_tr<5>), grid, block, 0, stream,
                       (const __bf16*)dY, (const __bf16*)X, dW, cs, pc);
    return;
  }
  if (tr_mode) {
    hipLaunchKernelGGL((conv_bwdw_kernel_tr<1>), grid, block, 0, stream,
                       (const __bf16*)dY, (const __bf16*)X, dW, cs, pc);
    return;
  }
  hipLaunchKernelGGL(conv_bwdw_kernel, grid, block, 0, stream,
                     (const __bf16*)dY, (const __bf16*)X, dW, ws, cs, pc);
}
