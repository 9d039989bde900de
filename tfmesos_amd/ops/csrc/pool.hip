// 3x3 stride-1 pad-1 average pooling, channels-last bf16 (the Inception
// block pool; torch's count_include_pad=True semantics -> divisor 9
// everywhere). The stencil is symmetric, so the SAME kernel computes
// the backward (dx = stencil(dy)): aten's NHWC avg_pool2d_backward
// measured 109 us avg vs ~15 us here (profiles/).
// Grid: one workgroup per (n, h) row; threads sweep (w, c) with bf16x8
// vector accesses when C % 8 == 0.
#include "common.h"

namespace {

template <bool VEC>
__global__ __launch_bounds__(256)
void avg3x3_kernel(const __bf16* __restrict__ X, __bf16* __restrict__ Y,
                   int N, int H, int W, int C) {
  const int nh = blockIdx.x;
  const int n = nh / H;
  const int h = nh - n * H;
  const long rowstride = (long)W * C;
  const __bf16* base = X + (long)n * H * rowstride;
  __bf16* out = Y + (long)n * H * rowstride + (long)h * rowstride;
  const __bf16* r0 = (h > 0) ? base + (long)(h - 1) * rowstride : nullptr;
  const __bf16* r1 = base + (long)h * rowstride;
  const __bf16* r2 = (h + 1 < H) ? base + (long)(h + 1) * rowstride : nullptr;
  const float inv9 = 1.f / 9.f;

  if (VEC) {
    const int WC = W * C;
    for (int i = ((int)blockIdx.y * 256 + threadIdx.x) * 8; i < WC;
         i += (int)gridDim.y * 256 * 8) {
      const int w = i / C;          // strip stays inside one w (C%8==0)
      const int c = i - w * C;
      f32x4 s0 = {}, s1 = {};
#pragma unroll
      for (int dw = -1; dw <= 1; ++dw) {
        const int ww = w + dw;
        if (ww < 0 || ww >= W) continue;
        const long off = (long)ww * C + c;
        const __bf16* rows[3] = {r0, r1, r2};
#pragma unroll
        for (int dr = 0; dr < 3; ++dr) {
          if (rows[dr] == nullptr) continue;
          bf16x8 v = *(const bf16x8*)(rows[dr] + off);
#pragma unroll
          for (int j = 0; j < 4; ++j) s0[j] += (float)v[j];
#pragma unroll
          for (int j = 0; j < 4; ++j) s1[j] += (float)v[4 + j];
        }
      }
      bf16x8 o;
#pragma unroll
      for (int j = 0; j < 4; ++j) o[j] = (__bf16)(s0[j] * inv9);
#pragma unroll
      for (int j = 0; j < 4; ++j) o[4 + j] = (__bf16)(s1[j] * inv9);
      *(bf16x8*)(out + (long)w * C + c) = o;
    }
  } else {
    const int WC = W * C;
    for (int i = (int)blockIdx.y * 256 + threadIdx.x; i < WC;
         i += (int)gridDim.y * 256) {
      const int w = i / C;
      const int c = i - w * C;
      float s = 0.f;
#pragma unroll
      for (int dw = -1; dw <= 1; ++dw) {
        const int ww = w + dw;
        if (ww < 0 || ww >= W) continue;
        const long off = (long)ww * C + c;
        if (r0) s += (float)r0[off];
        s += (float)r1[off];
        if (r2) s += (float)r2[off];
      }
      out[(long)w * C + c] = (__bf16)(s * inv9);
    }
  }
}

}  // namespace

void launch_avg3x3(const bf16_t* X, bf16_t* Y, int N, int H, int W, int C,
                   hipStream_t stream) {
  // split each (n,h) row over grid.y so small-spatial deep layers
  // (8^2 x 2048: N*H = 256 rows = 1 WG/CU) still fill the chip
  const long rows = (long)N * H;
  long want = (2048 + rows - 1) / rows;
  long per = ((long)W * C + 256 * 8 - 1) / (256 * 8);
  long ychunks = want < per ? want : per;
  if (ychunks < 1) ychunks = 1;
  dim3 grid((unsigned)rows, (unsigned)ychunks), block(256);
  if ((C & 7) == 0)
    hipLaunchKernelGGL((avg3x3_kernel<true>), grid, block, 0, stream,
                       (const __bf16*)X, (__bf16*)Y, N, H, W, C);
  else
    hipLaunchKernelGGL((avg3x3_kernel<false>), grid, block, 0, stream,
                       (const __bf16*)X, (__bf16*)Y, N, H, W, C);
}

namespace {

// 3x3 stride-2 max pool (no padding -> every window is fully in
// bounds), channels-last. Forward saves the winning tap (0..8, u8);
// backward GATHERS: each input pixel sums the dy of the <=2x2 windows
// whose saved tap points at it — deterministic, no atomics.
// ldo: output leading dim per pixel — a channel-narrow view of a wider
// channels-last tensor writes the Inception B/D reduction-block concat
// slice directly (no aten cat copy), mirroring the BN apply's strided
// store (csrc/bn.hip).
template <bool VEC>
__global__ __launch_bounds__(256)
void maxpool3x3s2_fwd_kernel(const __bf16* __restrict__ X,
                             __bf16* __restrict__ Y, long ldo,
                             unsigned char* __restrict__ idx,
                             int N, int H, int W, int C, int Ho, int Wo) {
  const int nho = blockIdx.x;
  const int n = nho / Ho;
  const int ho = nho - n * Ho;
  const __bf16* base = X + ((long)n * H + ho * 2) * W * C;
  __bf16* out = Y + ((long)n * Ho + ho) * Wo * ldo;
  unsigned char* oi = idx + ((long)n * Ho + ho) * Wo * C;
  const int WoC = Wo * C;
  const int step = VEC ? 8 : 1;
  for (int i = ((int)blockIdx.y * 256 + threadIdx.x) * step; i < WoC;
       i += (int)gridDim.y * 256 * step) {
    const int wo = i / C;
    const int c = i - wo * C;
    const long col0 = (long)wo * 2 * C + c;
    if (VEC) {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float best = -3.4e38f;
        int bi = 0;
#pragma unroll
        for (int r = 0; r < 3; ++r)
#pragma unroll
          for (int s = 0; s < 3; ++s) {
            const float v =
                (float)base[(long)r * W * C + col0 + (long)s * C + j];
            if (v > best) { best = v; bi = r * 3 + s; }
          }
        out[(long)wo * ldo + c + j] = (__bf16)best;
        oi[i + j] = (unsigned char)bi;
      }
    } else {
      float best = -3.4e38f;
      int bi = 0;
#pragma unroll
      for (int r = 0; r < 3; ++r)
#pragma unroll
        for (int s = 0; s < 3; ++s) {
          const float v = (float)base[(long)r * W * C + col0 + (long)s * C];
          if (v > best) { best = v; bi = r * 3 + s; }
        }
      out[(long)wo * ldo + c] = (__bf16)best;
      oi[i] = (unsigned char)bi;
    }
  }
}

typedef __attribute__((ext_vector_type(8))) unsigned char u8x8;

template <bool VEC>
__global__ __launch_bounds__(256)
void maxpool3x3s2_bwd_kernel(const __bf16* __restrict__ dY, long ldy,
                             const unsigned char* __restrict__ idx,
                             __bf16* __restrict__ dX,
                             int N, int H, int W, int C, int Ho, int Wo) {
  const int nh = blockIdx.x;
  const int n = nh / H;
  const int h = nh - n * H;
  const int WC = W * C;
  __bf16* out = dX + ((long)n * H + h) * WC;
  const int step = VEC ? 8 : 1;
  for (int i = ((int)blockIdx.y * 256 + threadIdx.x) * step; i < WC;
       i += (int)gridDim.y * 256 * step) {
    const int w = i / C;
    const int c = i - w * C;
    // windows (ho, wo) covering (h, w): ho*2 <= h <= ho*2+2
    const int ho_lo = (h - 2 + 1) / 2 < 0 ? 0 : (h - 2 + 1) / 2;
    const int ho_hi = h / 2 < Ho - 1 ? h / 2 : Ho - 1;
    const int wo_lo = (w - 2 + 1) / 2 < 0 ? 0 : (w - 2 + 1) / 2;
    const int wo_hi = w / 2 < Wo - 1 ? w / 2 : Wo - 1;
    if (VEC) {
      f32x4 a0 = {}, a1 = {};
      for (int ho = ho_lo; ho <= ho_hi; ++ho) {
        const int r = h - ho * 2;
        if (r < 0 || r > 2) continue;
        for (int wo = wo_lo; wo <= wo_hi; ++wo) {
          const int s = w - wo * 2;
          if (s < 0 || s > 2) continue;
          const long op = ((long)n * Ho + ho) * Wo + wo;
          const u8x8 iv = *(const u8x8*)&idx[op * C + c];
          const bf16x8 dv = *(const bf16x8*)&dY[op * ldy + c];
          const unsigned char want = (unsigned char)(r * 3 + s);
#pragma unroll
          for (int j = 0; j < 4; ++j)
            if (iv[j] == want) a0[j] += (float)dv[j];
#pragma unroll
          for (int j = 0; j < 4; ++j)
            if (iv[4 + j] == want) a1[j] += (float)dv[4 + j];
        }
      }
      bf16x8 o;
#pragma unroll
      for (int j = 0; j < 4; ++j) o[j] = (__bf16)a0[j];
#pragma unroll
      for (int j = 0; j < 4; ++j) o[4 + j] = (__bf16)a1[j];
      *(bf16x8*)(out + i) = o;
    } else {
      float acc = 0.f;
      for (int ho = ho_lo; ho <= ho_hi; ++ho) {
        const int r = h - ho * 2;
        if (r < 0 || r > 2) continue;
        for (int wo = wo_lo; wo <= wo_hi; ++wo) {
          const int s = w - wo * 2;
          if (s < 0 || s > 2) continue;
          const long op = ((long)n * Ho + ho) * Wo + wo;
          if (idx[op * C + c] == (unsigned char)(r * 3 + s))
            acc += (float)dY[op * ldy + c];
        }
      }
      out[i] = (__bf16)acc;
    }
  }
}

}  // namespace

void launch_maxpool3x3s2_fwd(const bf16_t* X, bf16_t* Y, long ldo,
                             unsigned char* idx,
                             int N, int H, int W, int C, int Ho, int Wo,
                             hipStream_t stream) {
  const long mrows = (long)N * Ho;
  long mwant = (2048 + mrows - 1) / mrows;
  long mper = ((long)Wo * C + 256L * 8 - 1) / (256 * 8);
  long mych = mwant < mper ? mwant : mper;
  if (mych < 1) mych = 1;
  dim3 grid((unsigned)mrows, (unsigned)mych), block(256);
  if ((C & 7) == 0)
    hipLaunchKernelGGL((maxpool3x3s2_fwd_kernel<true>), grid, block, 0,
                       stream, (const __bf16*)X, (__bf16*)Y, ldo, idx, N, H,
                       W, C, Ho, Wo);
  else
    hipLaunchKernelGGL((maxpool3x3s2_fwd_kernel<false>), grid, block, 0,
                       stream, (const __bf16*)X, (__bf16*)Y, ldo, idx, N, H,
                       W, C, Ho, Wo);
}

void launch_maxpool3x3s2_bwd(const bf16_t* dY, long ldy,
                             const unsigned char* idx,
                             bf16_t* dX, int N, int H, int W, int C, int Ho,
                             int Wo, hipStream_t stream) {
  const long mrows = (long)N * H;
  long mwant = (2048 + mrows - 1) / mrows;
  long mper = ((long)W * C + 256L * 8 - 1) / (256 * 8);
  long mych = mwant < mper ? mwant : mper;
  if (mych < 1) mych = 1;
  dim3 grid((unsigned)mrows, (unsigned)mych), block(256);
  if ((C & 7) == 0)
    hipLaunchKernelGGL((maxpool3x3s2_bwd_kernel<true>), grid, block, 0,
                       stream, (const __bf16*)dY, ldy, idx, (__bf16*)dX, N,
                       H, W, C, Ho, Wo);
  else
    hipLaunchKernelGGL((maxpool3x3s2_bwd_kernel<false>), grid, block, 0,
                       stream, (const __bf16*)dY, ldy, idx, (__bf16*)dX, N,
                       H, W, C, Ho, Wo);
}
