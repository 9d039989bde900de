// Fused optimizer apply kernels (PS side).
//
// The reference delegated parameter updates to TF's per-variable optimizer
// apply ops on the ps tasks (GradientDescent examples/mnist/mnist.py:55,
// Adam examples/mnist/mnist_replica.py:147). Here the WHOLE model lives in
// one flat fp32 master buffer (tfmesos_amd/ps/store.py), so each step is
// exactly ONE kernel: read grad, update master + optimizer state, and
// refresh the bf16 broadcast shadow in the same pass — one HBM round trip,
// float4-vectorized, grid sized to fill 256 CUs when the buffer is large.
#include "common.h"

namespace {

typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;

// grads arrive fp32 (collective-reduced) or bf16; templated load
template <typename G>
DEVINL float load_g(const G* g, long i);
template <>
DEVINL float load_g<float>(const float* g, long i) { return g[i]; }
template <>
DEVINL float load_g<bf16_t>(const bf16_t* g, long i) { return bf2f(g[i]); }

// vectorized 4-wide grad load (the flat buffers are 256-element aligned)
template <typename G>
DEVINL f32x4 load_g4(const G* g, long i);
template <>
DEVINL f32x4 load_g4<float>(const float* g, long i) {
  return *(const f32x4*)&g[i];
}
template <>
DEVINL f32x4 load_g4<bf16_t>(const bf16_t* g, long i) {
  const bf16x4 v = *(const bf16x4*)&g[i];
  f32x4 o;
#pragma unroll
  for (int j = 0; j < 4; ++j) o[j] = (float)v[j];
  return o;
}

template <typename G, bool MOM, bool BF16OUT>
__global__ void sgd_kernel(float* __restrict__ p, const G* __restrict__ g,
                           float* __restrict__ mbuf,
                           bf16_t* __restrict__ pbf, long n, float lr,
                           float momentum, float wd, float gscale,
                           float negdecay) {
  long i0 = (long)(blockIdx.x * blockDim.x + threadIdx.x) * 4;
  long stride = (long)gridDim.x * blockDim.x * 4;
  long i = i0;
  for (; i + 4 <= n; i += stride) {
    // b128 loads/stores on the aligned bulk (one HBM round trip for
    // master + grad + shadow)
    f32x4 gv = load_g4(g, i) * gscale;
    f32x4 pv = *(const f32x4*)&p[i];
    if (wd != 0.f) gv += wd * pv;
    if (negdecay != 0.f) {
      // soft nonnegativity penalty (NMF): g += c * min(p, 0) — folds
      // the per-factor clamp+add kernels into the apply pass
#pragma unroll
      for (int j = 0; j < 4; ++j)
        gv[j] += negdecay * fminf(pv[j], 0.f);
    }
    if (MOM) {
      f32x4 m = *(const f32x4*)&mbuf[i] * momentum + gv;
      *(f32x4*)&mbuf[i] = m;
      gv = m;
    }
    pv -= lr * gv;
    *(f32x4*)&p[i] = pv;
    if (BF16OUT) {
      bf16x4 o;
#pragma unroll
      for (int j = 0; j < 4; ++j) o[j] = (__bf16)pv[j];
      *(bf16x4*)&((__bf16*)pbf)[i] = o;
    }
  }
  for (; i < n; ++i) {      // tail (n % 4)
    float gv = load_g(g, i) * gscale;
    float pv = p[i];
    if (wd != 0.f) gv += wd * pv;
    if (negdecay != 0.f) gv += negdecay * fminf(pv, 0.f);
    if (MOM) {
      float m = mbuf[i] * momentum + gv;
      mbuf[i] = m;
      gv = m;
    }
    pv -= lr * gv;
    p[i] = pv;
    if (BF16OUT) pbf[i] = f2bf(pv);
  }
}

template <typename G, bool BF16OUT>
__global__ void adam_kernel(float* __restrict__ p, const G* __restrict__ g,
                            float* __restrict__ m, float* __restrict__ v,
                            bf16_t* __restrict__ pbf, long n, float lr,
                            float beta1, float beta2, float eps, float wd,
                            float bc1, float bc2, float gscale) {
  long i0 = (long)(blockIdx.x * blockDim.x + threadIdx.x) * 4;
  long stride = (long)gridDim.x * blockDim.x * 4;
  long i = i0;
  for (; i + 4 <= n; i += stride) {
    f32x4 gv = load_g4(g, i) * gscale;
    f32x4 pv = *(const f32x4*)&p[i];
    if (wd != 0.f) gv += wd * pv;
    f32x4 mv = beta1 * *(const f32x4*)&m[i] + (1.f - beta1) * gv;
    f32x4 vv = beta2 * *(const f32x4*)&v[i] + (1.f - beta2) * gv * gv;
    *(f32x4*)&m[i] = mv;
    *(f32x4*)&v[i] = vv;
#pragma unroll
    for (int j = 0; j < 4; ++j)
      pv[j] -= lr * (mv[j] / bc1) / (__builtin_sqrtf(vv[j] / bc2) + eps);
    *(f32x4*)&p[i] = pv;
    if (BF16OUT) {
      bf16x4 o;
#pragma unroll
      for (int j = 0; j < 4; ++j) o[j] = (__bf16)pv[j];
      *(bf16x4*)&((__bf16*)pbf)[i] = o;
    }
  }
  for (; i < n; ++i) {
    float gv = load_g(g, i) * gscale;
    float pv = p[i];
    if (wd != 0.f) gv += wd * pv;
    float mv = beta1 * m[i] + (1.f - beta1) * gv;
    float vv = beta2 * v[i] + (1.f - beta2) * gv * gv;
    m[i] = mv;
    v[i] = vv;
    pv -= lr * (mv / bc1) / (__builtin_sqrtf(vv / bc2) + eps);
    p[i] = pv;
    if (BF16OUT) pbf[i] = f2bf(pv);
  }
}

template <typename G, bool BF16OUT>
__global__ void adagrad_kernel(float* __restrict__ p, const G* __restrict__ g,
                               float* __restrict__ acc,
                               bf16_t* __restrict__ pbf, long n, float lr,
                               float eps, float wd, float gscale) {
  long i0 = (long)(blockIdx.x * blockDim.x + threadIdx.x) * 4;
  long stride = (long)gridDim.x * blockDim.x * 4;
  long i = i0;
  for (; i + 4 <= n; i += stride) {
    f32x4 gv = load_g4(g, i) * gscale;
    f32x4 pv = *(const f32x4*)&p[i];
    if (wd != 0.f) gv += wd * pv;
    f32x4 a = *(const f32x4*)&acc[i] + gv * gv;
    *(f32x4*)&acc[i] = a;
#pragma unroll
    for (int j = 0; j < 4; ++j)
      pv[j] -= lr * gv[j] / (__builtin_sqrtf(a[j]) + eps);
    *(f32x4*)&p[i] = pv;
    if (BF16OUT) {
      bf16x4 o;
#pragma unroll
      for (int j = 0; j < 4; ++j) o[j] = (__bf16)pv[j];
      *(bf16x4*)&((__bf16*)pbf)[i] = o;
    }
  }
  for (; i < n; ++i) {
    float gv = load_g(g, i) * gscale;
    float pv = p[i];
    if (wd != 0.f) gv += wd * pv;
    float a = acc[i] + gv * gv;
    acc[i] = a;
    pv -= lr * gv / (__builtin_sqrtf(a) + eps);
    p[i] = pv;
    if (BF16OUT) pbf[i] = f2bf(pv);
  }
}

inline dim3 apply_grid(long n) {
  // >=2048 workgroups fills 256 CUs x 8 blocks; small buffers get fewer
  long blocks = (n / 4 + 255) / 256;
  if (blocks > 4096) blocks = 4096;
  if (blocks < 1) blocks = 1;
  return dim3((unsigned)blocks);
}

}  // namespace

void launch_sgd(float* p, const void* g, bool g_bf16, float* mbuf,
                bf16_t* pbf, long n, float lr, float momentum, float wd,
                float gscale, float negdecay, hipStream_t stream) {
  dim3 grid = apply_grid(n), block(256);
#define DISP(GT, MOMV, OUTV)                                               \
  hipLaunchKernelGGL((sgd_kernel<GT, MOMV, OUTV>), grid, block, 0, stream, \
                     p, (const GT*)g, mbuf, pbf, n, lr, momentum, wd,     \
                     gscale, negdecay)
  if (g_bf16) {
    if (mbuf) { if (pbf) DISP(bf16_t, true, true); else DISP(bf16_t, true, false); }
    else      { if (pbf) DISP(bf16_t, false, true); else DISP(bf16_t, false, false); }
  } else {
    if (mbuf) { if (pbf) DISP(float, true, true); else DISP(float, true, false); }
    else      { if (pbf) DISP(float, false, true); else DISP(float, false, false); }
  }
#undef DISP
}

void launch_adam(float* p, const void* g, bool g_bf16, float* m, float* v,
                 bf16_t* pbf, long n, long step, float lr, float beta1,
                 float beta2, float eps, float wd, float gscale,
                 hipStream_t stream) {
  dim3 grid = apply_grid(n), block(256);
  float bc1 = 1.f - powf(beta1, (float)step);
  float bc2 = 1.f - powf(beta2, (float)step);
#define DISP(GT, OUTV)                                                      \
  hipLaunchKernelGGL((adam_kernel<GT, OUTV>), grid, block, 0, stream, p,    \
                     (const GT*)g, m, v, pbf, n, lr, beta1, beta2, eps, wd, \
                     bc1, bc2, gscale)
  if (g_bf16) { if (pbf) DISP(bf16_t, true); else DISP(bf16_t, false); }
  else        { if (pbf) DISP(float, true); else DISP(float, false); }
#undef DISP
}

void launch_adagrad(float* p, const void* g, bool g_bf16, float* acc,
                    bf16_t* pbf, long n, float lr, float eps, float wd,
                    float gscale, hipStream_t stream) {
  dim3 grid = apply_grid(n), block(256);
#define DISP(GT, OUTV)                                                    \
  hipLaunchKernelGGL((adagrad_kernel<GT, OUTV>), grid, block, 0, stream,  \
                     p, (const GT*)g, acc, pbf, n, lr, eps, wd, gscale)
  if (g_bf16) { if (pbf) DISP(bf16_t, true); else DISP(bf16_t, false); }
  else        { if (pbf) DISP(float, true); else DISP(float, false); }
#undef DISP
}
