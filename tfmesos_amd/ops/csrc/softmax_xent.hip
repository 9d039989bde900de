// Fused softmax + cross-entropy (fwd: mean loss + probs; bwd: dlogits).
//
// The reference workload's loss (softmax_cross_entropy_with_logits,
// examples/mnist/mnist_replica.py:143-145). One wave64 per row; the
// class dimension (10 for mnist, up to 4096 supported) is reduced with
// wave shuffles — no LDS round trip, no separate log-softmax pass.
#include "common.h"

namespace {

DEVINL float wave_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    v = fmaxf(v, __shfl_xor(v, off, WAVE));
  return v;
}

DEVINL float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    v += __shfl_xor(v, off, WAVE);
  return v;
}

// rows B, classes C; loss_out accumulates sum(-log p[label]) / B
__global__ void softmax_xent_fwd_kernel(const bf16_t* __restrict__ logits,
                                        const long* __restrict__ labels,
                                        bf16_t* __restrict__ probs,
                                        float* __restrict__ loss_out,
                                        int B, int C) {
  const int row = blockIdx.x;
  if (row >= B) return;
  const int lane = threadIdx.x;
  const bf16_t* lrow = logits + (long)row * C;
  bf16_t* prow = probs + (long)row * C;

  float mx = -1e30f;
  for (int c = lane; c < C; c += WAVE) mx = fmaxf(mx, bf2f(lrow[c]));
  mx = wave_max(mx);

  float sum = 0.f;
  for (int c = lane; c < C; c += WAVE) sum += __expf(bf2f(lrow[c]) - mx);
  sum = wave_sum(sum);
  const float inv = 1.f / sum;

  const long label = labels[row];
  float neglogp = 0.f;
  for (int c = lane; c < C; c += WAVE) {
    float p = __expf(bf2f(lrow[c]) - mx) * inv;
    prow[c] = f2bf(p);
    if (c == (int)label) neglogp = -__logf(fmaxf(p, 1e-30f));
  }
  neglogp = wave_sum(neglogp);
  if (lane == 0) atomicAdd(loss_out, neglogp / B);
}

// dlogits = (probs - onehot(label)) * scale
__global__ void softmax_xent_bwd_kernel(const bf16_t* __restrict__ probs,
                                        const long* __restrict__ labels,
                                        bf16_t* __restrict__ dlogits,
                                        float scale, int B, int C) {
  const int row = blockIdx.x;
  if (row >= B) return;
  const int lane = threadIdx.x;
  const bf16_t* prow = probs + (long)row * C;
  bf16_t* drow = dlogits + (long)row * C;
  const long label = labels[row];
  for (int c = lane; c < C; c += WAVE) {
    float v = bf2f(prow[c]) - (c == (int)label ? 1.f : 0.f);
    drow[c] = f2bf(v * scale);
  }
}

// Fully fused fwd+bwd for small class counts (mnist C=10): ONE kernel
// computes mean loss AND dlogits = (softmax - onehot)*scale, replacing
// the fwd kernel + bwd kernel + per-step loss zero-fill of the unfused
// path (each launch costs ~5us at this size — see profiles/). One
// workgroup, one row per thread (B<=256, C<=32): loss is tree-reduced in
// LDS and written once, so no atomics and no pre-zeroed buffer needed.
__global__ __launch_bounds__(256)
void softmax_xent_fused_kernel(const bf16_t* __restrict__ logits,
                               const long* __restrict__ labels,
                               bf16_t* __restrict__ dlogits,
                               float* __restrict__ loss_out, float scale,
                               int B, int C) {
  __shared__ float lsum[256];
  const int r = threadIdx.x;
  float neglogp = 0.f;
  if (r < B) {
    const bf16_t* lrow = logits + (long)r * C;
    bf16_t* drow = dlogits + (long)r * C;
    const int label = (int)labels[r];
    float mx = -1e30f;
    for (int c = 0; c < C; ++c) mx = fmaxf(mx, bf2f(lrow[c]));
    float sum = 0.f;
    for (int c = 0; c < C; ++c) sum += __expf(bf2f(lrow[c]) - mx);
    const float inv = 1.f / sum;
    for (int c = 0; c < C; ++c) {
      const float p = __expf(bf2f(lrow[c]) - mx) * inv;
      drow[c] = f2bf((p - (c == label ? 1.f : 0.f)) * scale);
      if (c == label) neglogp = -__logf(fmaxf(p, 1e-30f));
    }
  }
  lsum[r] = neglogp;
  __syncthreads();
#pragma unroll
  for (int s = 128; s > 0; s >>= 1) {
    if (r < s) lsum[r] += lsum[r + s];
    __syncthreads();
  }
  if (r == 0) *loss_out = lsum[0] / B;
}

}  // namespace

void launch_softmax_xent_fused(const bf16_t* logits, const long* labels,
                               bf16_t* dlogits, float* loss, float scale,
                               int B, int C, hipStream_t stream) {
  hipLaunchKernelGGL(softmax_xent_fused_kernel, dim3(1), dim3(256), 0, stream,
                     logits, labels, dlogits, loss, scale, B, C);
}

void launch_softmax_xent_fwd(const bf16_t* logits, const long* labels,
                             bf16_t* probs, float* loss, int B, int C,
                             hipStream_t stream) {
  hipLaunchKernelGGL(softmax_xent_fwd_kernel, dim3(B), dim3(WAVE), 0, stream,
                     logits, labels, probs, loss, B, C);
}

void launch_softmax_xent_bwd(const bf16_t* probs, const long* labels,
                             bf16_t* dlogits, float scale, int B, int C,
                             hipStream_t stream) {
  hipLaunchKernelGGL(softmax_xent_bwd_kernel, dim3(B), dim3(WAVE), 0, stream,
                     probs, labels, dlogits, scale, B, C);
}

namespace {

// Fully fused MLP classifier head for the mnist_replica hot step:
//   logits = h @ w + b ; softmax ; mean xent loss ;
//   dlogits = (p - onehot)/B ; dh = (dlogits @ w^T) * (h > 0)
// ONE single-workgroup kernel replacing three ~5-6 us launches (the
// per-kernel execution floor dominates at these sizes — see profiles/).
// One THREAD per row; h is staged through LDS with COALESCED
// cooperative copies (per-thread row walks over global memory are
// 64-way divergent and ran 10x slower), weights are read as f32x4
// broadcasts, dh is written back through the same LDS staging. Every
// class loop is unrolled to the 16-class cap: runtime-bounded loops
// over register arrays would demote acc[]/dl[] to scratch.
// Constraints: C <= 16, H <= 512, B <= 512, B*(H+8) <= 15000.
__global__ __launch_bounds__(512)
void mlp_head_fused_kernel(const bf16_t* __restrict__ h,
                           const bf16_t* __restrict__ w,
                           const bf16_t* __restrict__ bias,
                           const long* __restrict__ labels,
                           bf16_t* __restrict__ dlogits,
                           bf16_t* __restrict__ dh,
                           float* __restrict__ loss_out,
                           float scale, int B, int H, int C) {
  __shared__ __align__(16) float ws[512][16];    // 32 KB, f32x4 reads
  __shared__ __align__(16) __bf16 hs[15000];     // [B][H+8] staged rows
  __shared__ float lsum[512];
  const int t = threadIdx.x;
  const int hp = H + 8;
  for (int i = t; i < H * C; i += 512)
    ws[i / C][i % C] = bf2f(w[i]);
  for (int i = t; i < B * H; i += 512)
    hs[(i / H) * hp + (i % H)] = *(const __bf16*)&h[i];
  __syncthreads();

  float neglogp = 0.f;
  if (t < B) {
    __bf16* hrow = &hs[t * hp];
    float acc[16];
#pragma unroll
    for (int j = 0; j < 16; ++j) acc[j] = (j < C) ? bf2f(bias[j]) : 0.f;
    for (int e = 0; e < H; ++e) {
      const float hv = (float)hrow[e];
      const f32x4* wrow = (const f32x4*)ws[e];
#pragma unroll
      for (int q = 0; q < 4; ++q) {
        if (q * 4 >= C) break;
        const f32x4 wv = wrow[q];
#pragma unroll
        for (int j = 0; j < 4; ++j)
          if (q * 4 + j < C) acc[q * 4 + j] += hv * wv[j];
      }
    }
    float mx = -1e30f;
#pragma unroll
    for (int j = 0; j < 16; ++j) {
      if (j >= C) break;
      mx = fmaxf(mx, acc[j]);
    }
    float sum = 0.f;
#pragma unroll
    for (int j = 0; j < 16; ++j) {
      if (j >= C) break;
      acc[j] = __expf(acc[j] - mx);
      sum += acc[j];
    }
    const float inv = 1.f / sum;
    const int label = (int)labels[t];
    float dl[16];
    bf16_t* drow = dlogits + (long)t * C;
#pragma unroll
    for (int j = 0; j < 16; ++j) {
      if (j >= C) break;
      const float p = acc[j] * inv;
      dl[j] = (p - (j == label ? 1.f : 0.f)) * scale;
      drow[j] = f2bf(dl[j]);
      if (j == label) neglogp = -__logf(fmaxf(p, 1e-30f));
    }
    // dh into the LDS staging (overwrite h rows in place)
    for (int e = 0; e < H; ++e) {
      const float hv = (float)hrow[e];
      const f32x4* wrow = (const f32x4*)ws[e];
      float v = 0.f;
#pragma unroll
      for (int q = 0; q < 4; ++q) {
        if (q * 4 >= C) break;
        const f32x4 wv = wrow[q];
#pragma unroll
        for (int j = 0; j < 4; ++j)
          if (q * 4 + j < C) v += dl[q * 4 + j] * wv[j];
      }
      hrow[e] = (__bf16)(hv > 0.f ? v : 0.f);
    }
  }
  lsum[t] = neglogp;
  __syncthreads();
  // coalesced writeback of dh
  for (int i = t; i < B * H; i += 512)
    dh[i] = *(bf16_t*)&hs[(i / H) * hp + (i % H)];
#pragma unroll
  for (int s = 256; s > 0; s >>= 1) {
    if (t < s) lsum[t] += lsum[t + s];
    __syncthreads();
  }
  if (t == 0) *loss_out = lsum[0] / B;
}

}  // namespace

void launch_mlp_head_fused(const bf16_t* h, const bf16_t* w,
                           const bf16_t* bias, const long* labels,
                           bf16_t* dlogits, bf16_t* dh, float* loss,
                           float scale, int B, int H, int C,
                           hipStream_t stream) {
  hipLaunchKernelGGL(mlp_head_fused_kernel, dim3(1), dim3(512), 0, stream,
                     h, w, bias, labels, dlogits, dh, loss, scale, B, H, C);
}
