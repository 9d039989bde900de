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
