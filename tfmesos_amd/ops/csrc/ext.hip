// Python bindings for the tfmesos_amd CDNA4 kernels (torch extension).
// Pure dispatch + shape checking; all device code lives in the .hip files.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include "common.h"


// launchers from the .hip translation units
void launch_sgd(float*, const void*, bool, float*, bf16_t*, long, float,
                float, float, float, float, hipStream_t);
void launch_adam(float*, const void*, bool, float*, float*, bf16_t*, long,
                 long, float, float, float, float, float, float, hipStream_t);
void launch_adagrad(float*, const void*, bool, float*, bf16_t*, long, float,
                    float, float, float, hipStream_t);
void launch_gemm(const bf16_t*, const bf16_t*, const void*, bool, void*,
                 bool, const bf16_t*, void*, float*, int*, int, int, int,
                 int, int, int, int, int, bool, bool, int, int, int,
                 hipStream_t);
void launch_softmax_xent_fwd(const bf16_t*, const long*, bf16_t*, float*,
                             int, int, hipStream_t);
void launch_softmax_xent_fused(const bf16_t*, const long*, bf16_t*, float*,
                               float, int, int, hipStream_t);
void launch_softmax_xent_bwd(const bf16_t*, const long*, bf16_t*, float,
                             int, int, hipStream_t);
void launch_mlp_head_fused(const bf16_t*, const bf16_t*, const bf16_t*,
                           const long*, bf16_t*, bf16_t*, float*, void*,
                           void*, bool, float, int, int, int, hipStream_t);
void set_head_debug(void*);
void launch_mlp_fwd_head(const float*, int, const bf16_t*, const bf16_t*,
                         const bf16_t*, const long*, bf16_t*, bf16_t*,
                         float*, void*, void*, bool, float, int, int, int,
                         hipStream_t);
void launch_gemm_stripes(const bf16_t*, const bf16_t*, float*, int, int,
                         int, int, int, int, int, int, int, int,
                         hipStream_t);
void launch_gemm_small(const bf16_t*, const bf16_t*, const void*, bool,
                       void*, bool, int, void*, bool, int, int, int, int,
                       int, int, bool, const SmallSgdArgs*, hipStream_t);
void launch_gemm_stripes_any(const bf16_t*, const bf16_t*, float*, int, int,
                             int, int, int, int, int, int, bool, bool, int,
                             int, hipStream_t);
void launch_splitk_reduce_sgd(const float*, int, float*, bf16_t*, long,
                              float, float, float, hipStream_t);
void launch_gather_bf16(const bf16_t*, const long*, bf16_t*, long, int, long,
                        hipStream_t);
void launch_gather_f32(const float*, const long*, float*, long, int, long,
                       hipStream_t);
void launch_scatter_add_bf16(float*, const long*, const bf16_t*, long, int,
                             long, hipStream_t);
void launch_scatter_add_f32(float*, const long*, const float*, long, int,
                            long, hipStream_t);
void launch_relu_bwd(const bf16_t*, const bf16_t*, bf16_t*, long,
                     hipStream_t);
void launch_add_n(const bf16_t* const*, int, bf16_t*, long, hipStream_t);
void launch_colsum(const bf16_t*, float*, int, int, hipStream_t);
void launch_conv_fwd(const bf16_t*, const bf16_t*, const float*, bf16_t*,
                     float*, int, int, int, int, int, int, int, int, int,
                     int, int, int, int, bool, hipStream_t);
int conv_fwd_slices(int, int, int, int, int, int, int);
void launch_conv_bwd_data(const bf16_t*, long, const bf16_t*, bf16_t*,
                          float*,
                          int, int, int, int, int, int, int, int, int, int,
                          int, int, int, hipStream_t);
int conv_bwdd_slices(int, int, int, int, int, int, int);
void launch_conv_bwd_weight(const bf16_t*, long, const bf16_t*, float*,
                            int, int, int, int, int, int, int, int, int,
                            int, int, int, int, hipStream_t);
int conv_bwdw_slices(int, int, int, int, int, int, int);
void launch_bn_fwd(const bf16_t*, const bf16_t*, const bf16_t*, bf16_t*,
                   long, float*, float*, float*, long, int, int, float,
                   bool, hipStream_t);
void launch_bn_bwd(const bf16_t*, const bf16_t*, long, const bf16_t*,
                   const bf16_t*, const float*, const float*, bf16_t*,
                   bf16_t*, bf16_t*, float*, float*, float*, long, int,
                   int, bool, hipStream_t);
int bn_stats_slices(long, int);
int bn_max_channels();
int bn_group_slices(long, const int*, int);
void launch_bn_group_fwd(const bf16_t* const*, const bf16_t* const*,
                         const bf16_t* const*, const int*, int,
                         bf16_t* const*, const long*, float*, float*,
                         float*, long, int, float, bool, hipStream_t);
void launch_bn_group_bwd(const bf16_t* const*, const bf16_t* const*,
                         const long*,
                         const bf16_t* const*, const bf16_t* const*,
                         bf16_t* const*, const int*, int, const float*,
                         const float*, bf16_t*, bf16_t*, float*, float*,
                         float*, long, int, bool, hipStream_t);
void launch_avg3x3(const bf16_t*, bf16_t*, int, int, int, int, hipStream_t);
void launch_maxpool3x3s2_fwd(const bf16_t*, bf16_t*, long, unsigned char*, int,
                             int, int, int, int, int, hipStream_t);
void launch_maxpool3x3s2_bwd(const bf16_t*, long, const unsigned char*,
                             bf16_t*, int, int, int, int, int, int,
                             hipStream_t);

namespace {

hipStream_t cur_stream() {
  return at::cuda::getCurrentCUDAStream().stream();
}

const void* grad_ptr(const torch::Tensor& g, bool* is_bf16) {
  TORCH_CHECK(g.is_contiguous(), "grad must be contiguous");
  if (g.scalar_type() == torch::kBFloat16) {
    *is_bf16 = true;
    return g.data_ptr();
  }
  TORCH_CHECK(g.scalar_type() == torch::kFloat32, "grad must be fp32 or bf16");
  *is_bf16 = false;
  return g.data_ptr();
}

float* opt_f32(torch::Tensor& t, long n, const char* name) {
  if (t.numel() == 0) return nullptr;
  TORCH_CHECK(t.numel() == n && t.scalar_type() == torch::kFloat32,
              name, " must be fp32 with same numel as param");
  return t.data_ptr<float>();
}

bf16_t* opt_bf16(torch::Tensor& t, long n, const char* name) {
  if (t.numel() == 0) return nullptr;
  TORCH_CHECK(t.numel() == n && t.scalar_type() == torch::kBFloat16,
              name, " must be bf16 with same numel as param");
  return (bf16_t*)t.data_ptr();
}

void fused_sgd(torch::Tensor param, torch::Tensor grad,
               torch::Tensor momentum_buf, torch::Tensor bf16_out, double lr,
               double momentum, double weight_decay, double grad_scale,
               double neg_decay) {
  TORCH_CHECK(param.is_cuda() && param.is_contiguous() &&
              param.scalar_type() == torch::kFloat32,
              "param must be contiguous fp32 on GPU");
  long n = param.numel();
  TORCH_CHECK(grad.numel() == n, "grad/param numel mismatch");
  bool gb;
  const void* g = grad_ptr(grad, &gb);
  launch_sgd(param.data_ptr<float>(), g, gb,
             opt_f32(momentum_buf, n, "momentum_buf"),
             opt_bf16(bf16_out, n, "bf16_out"), n, (float)lr,
             (float)momentum, (float)weight_decay, (float)grad_scale,
             (float)neg_decay, cur_stream());
}

void fused_adam(torch::Tensor param, torch::Tensor grad, torch::Tensor m,
                torch::Tensor v, torch::Tensor bf16_out, long step, double lr,
                double beta1, double beta2, double eps, double weight_decay,
                double grad_scale) {
  long n = param.numel();
  TORCH_CHECK(param.is_cuda() && param.scalar_type() == torch::kFloat32 &&
              grad.numel() == n && m.numel() == n && v.numel() == n,
              "adam tensor mismatch");
  bool gb;
  const void* g = grad_ptr(grad, &gb);
  launch_adam(param.data_ptr<float>(), g, gb, m.data_ptr<float>(),
              v.data_ptr<float>(), opt_bf16(bf16_out, n, "bf16_out"), n, step,
              (float)lr, (float)beta1, (float)beta2, (float)eps,
              (float)weight_decay, (float)grad_scale, cur_stream());
}

void fused_adagrad(torch::Tensor param, torch::Tensor grad,
                   torch::Tensor accum, torch::Tensor bf16_out, double lr,
                   double eps, double weight_decay, double grad_scale) {
  long n = param.numel();
  TORCH_CHECK(param.is_cuda() && param.scalar_type() == torch::kFloat32 &&
              grad.numel() == n && accum.numel() == n,
              "adagrad tensor mismatch");
  bool gb;
  const void* g = grad_ptr(grad, &gb);
  launch_adagrad(param.data_ptr<float>(), g, gb, accum.data_ptr<float>(),
                 opt_bf16(bf16_out, n, "bf16_out"), n, (float)lr, (float)eps,
                 (float)weight_decay, (float)grad_scale, cur_stream());
}

// staging vector width for an operand: b128 when every row start stays
// 16B-aligned, b32 when 4B, else scalar
int vec_level(const void* base, long ld) {
  if (((uintptr_t)base % 16 == 0) && (ld % 8 == 0)) return 8;
  if (((uintptr_t)base % 4 == 0) && (ld % 2 == 0)) return 2;
  return 1;
}

// persistent per-device split-K workspace (one fp32 stripe per K-slice,
// fully overwritten by every launch) + tile arrival counters (the
// last-arriver epilogue resets them, so they stay zeroed between
// stream-ordered launches).
float* splitk_ws(const torch::Device& dev, long n, long ntiles, int** cnt) {
  static std::unordered_map<int, torch::Tensor> wsmap, cntmap;
  const int idx = dev.index();
  auto& w = wsmap[idx];
  if (!w.defined() || w.numel() < n)
    w = torch::zeros({n}, torch::dtype(torch::kFloat32).device(dev));
  auto& c = cntmap[idx];
  if (!c.defined() || c.numel() < ntiles)
    c = torch::zeros({ntiles}, torch::dtype(torch::kInt32).device(dev));
  *cnt = c.data_ptr<int>();
  return w.data_ptr<float>();
}

torch::Tensor gemm_bias_act_out(torch::Tensor a, torch::Tensor b,
                                torch::Tensor bias, long act, bool trans_a,
                                bool trans_b, torch::Tensor out,
                                torch::Tensor aux, torch::Tensor colsum_out) {
  TORCH_CHECK(a.is_cuda() && b.is_cuda(), "gemm: tensors must be on GPU");
  TORCH_CHECK(a.scalar_type() == torch::kBFloat16 &&
              b.scalar_type() == torch::kBFloat16, "gemm: bf16 inputs only");
  TORCH_CHECK(a.is_contiguous() && b.is_contiguous() && a.dim() == 2 &&
              b.dim() == 2, "gemm: contiguous 2-D inputs");
  int M = trans_a ? a.size(1) : a.size(0);
  int Ka = trans_a ? a.size(0) : a.size(1);
  int Kb = trans_b ? b.size(1) : b.size(0);
  int N = trans_b ? b.size(0) : b.size(1);
  TORCH_CHECK(Ka == Kb, "gemm: inner dims mismatch ", Ka, " vs ", Kb);
  const void* bias_p = nullptr;
  bool bias_bf16 = false;
  if (bias.numel() > 0) {
    TORCH_CHECK((bias.scalar_type() == torch::kFloat32 ||
                 bias.scalar_type() == torch::kBFloat16) && bias.numel() == N,
                "bias must be fp32 or bf16 [N]");
    bias_bf16 = bias.scalar_type() == torch::kBFloat16;
    bias_p = bias.data_ptr();
  }
  bool out_f32;
  if (out.numel() == 0) {
    out = torch::empty({M, N}, a.options());
    out_f32 = false;
  } else {
    TORCH_CHECK(out.is_contiguous() && out.dim() == 2 && out.size(0) == M &&
                out.size(1) == N, "out shape mismatch");
    out_f32 = out.scalar_type() == torch::kFloat32;
    TORCH_CHECK(out_f32 || out.scalar_type() == torch::kBFloat16,
                "out must be fp32 or bf16");
  }
  const bf16_t* aux_p = nullptr;
  if (act == 2) {
    TORCH_CHECK(!trans_a && trans_b && !out_f32 && bias_p == nullptr,
                "act=relu_bwd needs nt, bf16 out, no bias");
    TORCH_CHECK(aux.scalar_type() == torch::kBFloat16 && aux.is_contiguous()
                && aux.numel() == (long)M * N, "aux must be bf16 [M,N]");
    aux_p = (const bf16_t*)aux.data_ptr();
  } else if (act == 3) {
    // out = A@B - aux (the NMF residual, fused into the epilogue)
    TORCH_CHECK(!trans_a && !trans_b && !out_f32 && bias_p == nullptr,
                "act=sub needs nn, bf16 out, no bias");
    TORCH_CHECK(aux.scalar_type() == torch::kBFloat16 && aux.is_contiguous()
                && aux.numel() == (long)M * N, "aux must be bf16 [M,N]");
    aux_p = (const bf16_t*)aux.data_ptr();
  }
  void* colsum_p = nullptr;
  if (colsum_out.numel() > 0) {
    TORCH_CHECK(trans_a && !trans_b && bias_p == nullptr && act == 0,
                "colsum fusion needs tn, no bias, no act");
    TORCH_CHECK(colsum_out.is_contiguous() && colsum_out.numel() == N &&
                colsum_out.scalar_type() == out.scalar_type(),
                "colsum_out must match out dtype, [N]");
    colsum_p = colsum_out.data_ptr();
  }
  // small-tile ONE-kernel path for tiny tile grids (mnist fwd
  // 100x100x784 and dW1 784x100x100): the 4 waves split K in-block
  // (no split-K reduce pass — the second dispatch floor cost more
  // than the whole GEMM at these sizes)
  // gate: enough 32x32 tiles to feed the memory system (at 16 blocks
  // the one-kernel form measured 16.5 us vs 11.1 for split-K + reduce
  // on the mnist fwd shape — parallelism beats the saved dispatch
  // floor there), few enough that the shape is in the tiny-GEMM
  // regime, and K short enough that a wave's quarter is 1-2 chunks
  const long stiles = (long)((M + 31) / 32) * ((N + 31) / 32);
  const bool cs_small_ok =
      colsum_p == nullptr ||
      (trans_a && !trans_b && bias_p == nullptr && act == 0 &&
       colsum_out.scalar_type() == out.scalar_type());
  if (act <= 1 && !trans_b && aux_p == nullptr && cs_small_ok &&
      stiles >= 32 && stiles <= 512 && Ka <= 256) {
    launch_gemm_small((const bf16_t*)a.data_ptr(),
                      (const bf16_t*)b.data_ptr(), bias_p, bias_bf16,
                      out.data_ptr(), out_f32, act == 1 ? 1 : 0, colsum_p,
                      out_f32, M, N, Ka, a.size(1), b.size(1), N, trans_a,
                      nullptr, cur_stream());
    return out;
  }
  // split-K when the plain tile grid can't feed the 256-CU chip and K
  // has enough depth to slice (any transpose combo; not with the
  // fused colsum/relu_bwd epilogues — those stay single-phase)
  const int nx = (N + 63) / 64, ny = (M + 63) / 64;
  int nslice = 1, kc = 0;
  float* ws = nullptr;
  int* cnt = nullptr;
  // colsum-fused GEMMs (tn, no bias/act) can slice too: partials go to
  // an extra [nslice, N] stripe area reduced in phase 2. Same K gate as
  // plain shapes — slicing SHALLOW colsum GEMMs (mnist dW1, K=100)
  // measured slower: the second (reduce) launch costs more than the
  // occupancy it buys.
  const bool cs_ok = colsum_p == nullptr ||
                     (trans_a && !trans_b && bias_p == nullptr && act == 0);
  const long kmin = 256;
  if (act <= 1 && cs_ok && nx * ny < 256 && Ka >= kmin) {
    int want = std::min(colsum_p != nullptr ? (int)((Ka + 31) / 32)
                                            : (int)(Ka / 64),
                        256 / (nx * ny));
    if (want > 16) want = 16;
    static int want_env = -2;
    if (want_env == -2) {
      const char* e = getenv("TFA_SK_WANT");   // tuning override
      want_env = e ? atoi(e) : -1;
    }
    if (want_env > 0) want = want_env;
    if (want > 1) {
      kc = ((Ka + want - 1) / want + 31) / 32 * 32;
      nslice = (Ka + kc - 1) / kc;
      if (nslice > 1)
        ws = splitk_ws(a.device(),
                       (long)M * N * nslice +
                           (colsum_p != nullptr ? (long)nslice * N : 0),
                       (long)nx * ny, &cnt);
    }
  }
  launch_gemm((const bf16_t*)a.data_ptr(), (const bf16_t*)b.data_ptr(),
              bias_p, bias_bf16, out.data_ptr(), out_f32, aux_p, colsum_p,
              ws, cnt, kc, nslice, M, N, Ka, a.size(1), b.size(1), N,
              trans_a, trans_b, (int)act,
              vec_level(a.data_ptr(), a.size(1)),
              vec_level(b.data_ptr(), b.size(1)), cur_stream());
  return out;
}

// GEMM -> SGD fusion for the NMF factor updates: the gradient
// G = op(A)@op(B) never materializes — the split-K stripes feed
// p -= lr*(gscale*G + nd*min(p,0)) and the bf16 shadow refresh in the
// phase-2 reduce kernel (saves the separate apply launch AND the
// gradient round trip). Falls back to plain GEMM + fused_sgd when
// split-K does not engage at this shape.
void gemm_sgd(torch::Tensor a, torch::Tensor b, bool trans_a, bool trans_b,
              torch::Tensor param, torch::Tensor shadow, double lr,
              double grad_scale, double neg_decay) {
  TORCH_CHECK(a.is_cuda() && b.is_cuda() && a.is_contiguous() &&
              b.is_contiguous() && a.dim() == 2 && b.dim() == 2 &&
              a.scalar_type() == torch::kBFloat16 &&
              b.scalar_type() == torch::kBFloat16, "gemm_sgd: bf16 2-D");
  const int M = trans_a ? a.size(1) : a.size(0);
  const int Ka = trans_a ? a.size(0) : a.size(1);
  const int Kb = trans_b ? b.size(1) : b.size(0);
  const int N = trans_b ? b.size(0) : b.size(1);
  TORCH_CHECK(Ka == Kb, "gemm_sgd: inner dims mismatch");
  TORCH_CHECK(param.is_contiguous() &&
              param.scalar_type() == torch::kFloat32 &&
              param.numel() == (long)M * N, "param must be fp32 [M*N]");
  bf16_t* shp = nullptr;
  if (shadow.numel() > 0) {
    TORCH_CHECK(shadow.is_contiguous() &&
                shadow.scalar_type() == torch::kBFloat16 &&
                shadow.numel() == (long)M * N, "shadow must be bf16 [M*N]");
    shp = (bf16_t*)shadow.data_ptr();
  }
  const int nx = (N + 63) / 64, ny = (M + 63) / 64;
  int nslice = 1, kc = 0;
  if (nx * ny < 256 && Ka >= 256) {
    int want = std::min((int)(Ka / 64), 256 / (nx * ny));
    if (want > 16) want = 16;
    if (want > 1) {
      kc = ((Ka + want - 1) / want + 31) / 32 * 32;
      nslice = (Ka + kc - 1) / kc;
    }
  }
  if (nslice > 1) {
    int* cnt_unused;
    float* ws = splitk_ws(a.device(), (long)M * N * nslice,
                          (long)nx * ny, &cnt_unused);
    launch_gemm_stripes_any((const bf16_t*)a.data_ptr(),
                            (const bf16_t*)b.data_ptr(), ws, kc, nslice,
                            M, N, Ka, a.size(1), b.size(1), N, trans_a,
                            trans_b, vec_level(a.data_ptr(), a.size(1)),
                            vec_level(b.data_ptr(), b.size(1)),
                            cur_stream());
    launch_splitk_reduce_sgd(ws, nslice, param.data_ptr<float>(), shp,
                             (long)M * N, (float)lr, (float)grad_scale,
                             (float)neg_decay, cur_stream());
    return;
  }
  // no split-K at this shape: grad to a temp, then the fused apply
  auto gtmp = torch::empty({M, N}, a.options().dtype(torch::kFloat32));
  launch_gemm((const bf16_t*)a.data_ptr(), (const bf16_t*)b.data_ptr(),
              nullptr, false, gtmp.data_ptr(), true, nullptr, nullptr,
              nullptr, nullptr, 0, 1, M, N, Ka, a.size(1), b.size(1), N,
              trans_a, trans_b, 0, vec_level(a.data_ptr(), a.size(1)),
              vec_level(b.data_ptr(), b.size(1)), cur_stream());
  launch_sgd(param.data_ptr<float>(), gtmp.data_ptr(), false, nullptr, shp,
             (long)M * N, (float)lr, 0.f, 0.f, (float)grad_scale,
             (float)neg_decay, cur_stream());
}

// Paired form preserving SIMULTANEOUS-update semantics (the reference
// applies all gradients of a step at once): BOTH factors' stripe
// phases launch before EITHER fused apply, reading only pre-update
// values. One shared workspace allocation covers both.
void gemm_sgd_pair(torch::Tensor a1, torch::Tensor b1, bool ta1, bool tb1,
                   torch::Tensor p1, torch::Tensor s1,
                   torch::Tensor a2, torch::Tensor b2, bool ta2, bool tb2,
                   torch::Tensor p2, torch::Tensor s2,
                   double lr, double grad_scale, double neg_decay) {
  struct Half {
    torch::Tensor a, b, p, s;
    bool ta, tb;
    int M, N, K, nslice, kc;
    long ws_off;
    torch::Tensor gtmp;
  } h[2] = {{a1, b1, p1, s1, ta1, tb1}, {a2, b2, p2, s2, ta2, tb2}};
  long ws_total = 0;
  for (int i = 0; i < 2; ++i) {
    auto& x = h[i];
    x.M = x.ta ? x.a.size(1) : x.a.size(0);
    x.K = x.ta ? x.a.size(0) : x.a.size(1);
    x.N = x.tb ? x.b.size(0) : x.b.size(1);
    TORCH_CHECK(x.p.is_contiguous() &&
                x.p.scalar_type() == torch::kFloat32 &&
                x.p.numel() == (long)x.M * x.N, "gemm_sgd_pair: bad param");
    const int nx = (x.N + 63) / 64, ny = (x.M + 63) / 64;
    x.nslice = 1;
    x.kc = 0;
    if (nx * ny < 256 && x.K >= 256) {
      int want = std::min((int)(x.K / 64), 256 / (nx * ny));
      if (want > 16) want = 16;
      if (want > 1) {
        x.kc = ((x.K + want - 1) / want + 31) / 32 * 32;
        x.nslice = (x.K + x.kc - 1) / x.kc;
      }
    }
    x.ws_off = ws_total;
    if (x.nslice > 1) ws_total += (long)x.M * x.N * x.nslice;
  }
  torch::Tensor ws;
  if (ws_total > 0)
    ws = torch::empty({ws_total}, a1.options().dtype(torch::kFloat32));
  // phase 1 for BOTH halves (reads only pre-update operands)
  for (int i = 0; i < 2; ++i) {
    auto& x = h[i];
    if (x.nslice > 1) {
      launch_gemm_stripes_any((const bf16_t*)x.a.data_ptr(),
                              (const bf16_t*)x.b.data_ptr(),
                              ws.data_ptr<float>() + x.ws_off, x.kc,
                              x.nslice, x.M, x.N, x.K, x.a.size(1),
                              x.b.size(1), x.N,
                              x.ta, x.tb,
                              vec_level(x.a.data_ptr(), x.a.size(1)),
                              vec_level(x.b.data_ptr(), x.b.size(1)),
                              cur_stream());
    } else {
      x.gtmp = torch::empty({x.M, x.N},
                            a1.options().dtype(torch::kFloat32));
      launch_gemm((const bf16_t*)x.a.data_ptr(),
                  (const bf16_t*)x.b.data_ptr(), nullptr, false,
                  x.gtmp.data_ptr(), true, nullptr, nullptr, nullptr,
                  nullptr, 0, 1, x.M, x.N, x.K, x.a.size(1), x.b.size(1),
                  x.N, x.ta, x.tb, 0,
                  vec_level(x.a.data_ptr(), x.a.size(1)),
                  vec_level(x.b.data_ptr(), x.b.size(1)), cur_stream());
    }
  }
  // phase 2: fused applies
  for (int i = 0; i < 2; ++i) {
    auto& x = h[i];
    bf16_t* shp = x.s.numel() ? (bf16_t*)x.s.data_ptr() : nullptr;
    if (x.nslice > 1) {
      launch_splitk_reduce_sgd(ws.data_ptr<float>() + x.ws_off, x.nslice,
                               x.p.data_ptr<float>(), shp,
                               (long)x.M * x.N, (float)lr,
                               (float)grad_scale, (float)neg_decay,
                               cur_stream());
    } else {
      launch_sgd(x.p.data_ptr<float>(), x.gtmp.data_ptr(), false, nullptr,
                 shp, (long)x.M * x.N, (float)lr, 0.f, 0.f,
                 (float)grad_scale, (float)neg_decay, cur_stream());
    }
  }
}

torch::Tensor add_n(std::vector<torch::Tensor> xs) {
  const int n = (int)xs.size();
  TORCH_CHECK(n >= 1 && n <= 6, "add_n: 1..6 tensors");
  const long total = xs[0].numel();
  const bf16_t* srcs[6];
  for (int i = 0; i < n; ++i) {
    // dense in EITHER memory format (the Python wrapper makes all
    // inputs the same layout, so flat elementwise math is valid)
    TORCH_CHECK(xs[i].is_cuda() &&
                (xs[i].is_contiguous() ||
                 xs[i].is_contiguous(torch::MemoryFormat::ChannelsLast)) &&
                xs[i].scalar_type() == torch::kBFloat16 &&
                xs[i].numel() == total, "add_n: same-shape dense bf16");
    srcs[i] = (const bf16_t*)xs[i].data_ptr();
  }
  auto out = torch::empty_like(xs[0]);
  launch_add_n(srcs, n, (bf16_t*)out.data_ptr(), total, cur_stream());
  return out;
}

torch::Tensor gemm_bias_act(torch::Tensor a, torch::Tensor b,
                            torch::Tensor bias, long act, bool trans_a,
                            bool trans_b) {
  auto empty = torch::empty({0}, a.options());
  return gemm_bias_act_out(a, b, bias, act, trans_a, trans_b, empty, empty,
                           empty);
}

std::tuple<torch::Tensor, torch::Tensor> softmax_xent_fwd(
    torch::Tensor logits, torch::Tensor labels) {
  TORCH_CHECK(logits.is_cuda() && logits.scalar_type() == torch::kBFloat16 &&
              logits.is_contiguous() && logits.dim() == 2,
              "logits must be contiguous bf16 [B,C] on GPU");
  TORCH_CHECK(labels.scalar_type() == torch::kInt64 &&
              labels.numel() == logits.size(0), "labels must be i64 [B]");
  int B = logits.size(0), C = logits.size(1);
  auto probs = torch::empty_like(logits);
  auto loss = torch::zeros({}, logits.options().dtype(torch::kFloat32));
  launch_softmax_xent_fwd((const bf16_t*)logits.data_ptr(),
                          labels.data_ptr<long>(), (bf16_t*)probs.data_ptr(),
                          loss.data_ptr<float>(), B, C, cur_stream());
  return {loss, probs};
}

// fully fused fwd+bwd: returns (mean loss, dlogits=(softmax-onehot)*scale)
// in ONE kernel for small shapes; composes the two-kernel path otherwise
std::tuple<torch::Tensor, torch::Tensor> softmax_xent_fused(
    torch::Tensor logits, torch::Tensor labels, double scale) {
  TORCH_CHECK(logits.is_cuda() && logits.scalar_type() == torch::kBFloat16 &&
              logits.is_contiguous() && logits.dim() == 2,
              "logits must be contiguous bf16 [B,C] on GPU");
  TORCH_CHECK(labels.scalar_type() == torch::kInt64 &&
              labels.numel() == logits.size(0), "labels must be i64 [B]");
  int B = logits.size(0), C = logits.size(1);
  auto dlogits = torch::empty_like(logits);
  if (B <= 256 && C <= 32) {
    auto loss = torch::empty({}, logits.options().dtype(torch::kFloat32));
    launch_softmax_xent_fused((const bf16_t*)logits.data_ptr(),
                              labels.data_ptr<long>(),
                              (bf16_t*)dlogits.data_ptr(),
                              loss.data_ptr<float>(), (float)scale, B, C,
                              cur_stream());
    return {loss, dlogits};
  }
  auto loss = torch::zeros({}, logits.options().dtype(torch::kFloat32));
  auto probs = torch::empty_like(logits);
  launch_softmax_xent_fwd((const bf16_t*)logits.data_ptr(),
                          labels.data_ptr<long>(), (bf16_t*)probs.data_ptr(),
                          loss.data_ptr<float>(), B, C, cur_stream());
  launch_softmax_xent_bwd((const bf16_t*)probs.data_ptr(),
                          labels.data_ptr<long>(),
                          (bf16_t*)dlogits.data_ptr(), (float)scale, B, C,
                          cur_stream());
  return {loss, dlogits};
}

// fused classifier head: logits=h@w+b, softmax-xent, dlogits, and
// dh = (dlogits @ w^T) * (h>0), all in ONE single-workgroup kernel
// dw2/db2 (optional, pass empty to skip): classifier weight/bias grads
// computed inside the fused kernel (fp32 or bf16 — the flat grad
// buffer's dtype). Only supported on the MFMA path (B<=128, H<=128).
std::vector<torch::Tensor> mlp_head_fused(torch::Tensor h, torch::Tensor w,
                                          torch::Tensor b, torch::Tensor labels,
                                          double scale, torch::Tensor dw2,
                                          torch::Tensor db2) {
  TORCH_CHECK(h.is_cuda() && h.dim() == 2 && h.is_contiguous() &&
              h.scalar_type() == torch::kBFloat16, "h must be bf16 [B,H]");
  TORCH_CHECK(w.is_contiguous() && w.dim() == 2 && w.size(0) == h.size(1) &&
              w.scalar_type() == torch::kBFloat16, "w must be bf16 [H,C]");
  const int B = h.size(0), H = h.size(1), C = w.size(1);
  TORCH_CHECK(C <= 16 && H <= 512 && B <= 512 &&
              (long)B * (H + 8) <= 15000,
              "mlp_head_fused limits: C<=16, H<=512, B<=512, "
              "B*(H+8)<=15000");
  TORCH_CHECK(b.scalar_type() == torch::kBFloat16 && b.numel() == C,
              "bias must be bf16 [C]");
  TORCH_CHECK(labels.scalar_type() == torch::kInt64 && labels.numel() == B);
  auto dlogits = torch::empty({B, C}, h.options());
  auto dh = torch::empty_like(h);
  auto loss = torch::empty({}, h.options().dtype(torch::kFloat32));
  void* dw2p = nullptr;
  void* db2p = nullptr;
  bool gf32 = true;
  if (dw2.numel() > 0) {
    TORCH_CHECK(B <= 128 && H <= 128,
                "fused dw2 requires the MFMA head path (B<=128, H<=128)");
    TORCH_CHECK(dw2.is_contiguous() && dw2.numel() == (long)H * C &&
                db2.is_contiguous() && db2.numel() == C &&
                dw2.scalar_type() == db2.scalar_type() &&
                (dw2.scalar_type() == torch::kFloat32 ||
                 dw2.scalar_type() == torch::kBFloat16),
                "dw2/db2 must be contiguous fp32 or bf16 [H*C]/[C]");
    gf32 = dw2.scalar_type() == torch::kFloat32;
    dw2p = dw2.data_ptr();
    db2p = db2.data_ptr();
  }
  launch_mlp_head_fused((const bf16_t*)h.data_ptr(),
                        (const bf16_t*)w.data_ptr(),
                        (const bf16_t*)b.data_ptr(), labels.data_ptr<long>(),
                        (bf16_t*)dlogits.data_ptr(), (bf16_t*)dh.data_ptr(),
                        loss.data_ptr<float>(), dw2p, db2p, gf32,
                        (float)scale, B, H, C, cur_stream());
  return {loss, dlogits, dh};
}

// Whole mnist fwd+head in TWO kernels: split-K GEMM stripes for
// h = relu(x @ w1 + b1), then the fused head consumes the stripes
// directly (h never exists in global memory). Returns (loss, dh);
// dw2/db2 grads are written by the head. Falls back to a materialized
// h + the normal head when split-K does not engage.
std::vector<torch::Tensor> mlp_fwd_head_fused(
    torch::Tensor x, torch::Tensor w1, torch::Tensor b1, torch::Tensor w2,
    torch::Tensor b2, torch::Tensor labels, double scale,
    torch::Tensor dw2, torch::Tensor db2) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 && x.is_contiguous() &&
              x.scalar_type() == torch::kBFloat16, "x must be bf16 [B,K]");
  TORCH_CHECK(w1.is_contiguous() && w1.dim() == 2 &&
              w1.size(0) == x.size(1) &&
              w1.scalar_type() == torch::kBFloat16, "w1 must be bf16 [K,H]");
  const int B = x.size(0), K1 = x.size(1), H = w1.size(1);
  const int C = w2.size(1);
  TORCH_CHECK(B <= 128 && H <= 128 && (H & 3) == 0 && C <= 16,
              "mlp_fwd_head_fused limits: B<=128, H<=128 (H%4==0), C<=16");
  TORCH_CHECK(b1.scalar_type() == torch::kBFloat16 && b1.numel() == H);
  TORCH_CHECK(w2.is_contiguous() && w2.size(0) == H &&
              w2.scalar_type() == torch::kBFloat16, "w2 must be bf16 [H,C]");
  TORCH_CHECK(b2.scalar_type() == torch::kBFloat16 && b2.numel() == C);
  TORCH_CHECK(labels.scalar_type() == torch::kInt64 && labels.numel() == B);
  auto dlogits = torch::empty({B, C}, x.options());
  auto dh = torch::empty({B, H}, x.options());
  auto loss = torch::empty({}, x.options().dtype(torch::kFloat32));
  void* dw2p = nullptr;
  void* db2p = nullptr;
  bool gf32 = true;
  if (dw2.numel() > 0) {
    TORCH_CHECK(dw2.is_contiguous() && dw2.numel() == (long)H * C &&
                db2.is_contiguous() && db2.numel() == C &&
                dw2.scalar_type() == db2.scalar_type() &&
                (dw2.scalar_type() == torch::kFloat32 ||
                 dw2.scalar_type() == torch::kBFloat16),
                "dw2/db2 must be contiguous fp32 or bf16");
    gf32 = dw2.scalar_type() == torch::kFloat32;
    dw2p = dw2.data_ptr();
    db2p = db2.data_ptr();
  }
  // split-K sliced LOW (default 4, TFA_HEAD_NSLICE to tune): the head
  // workgroups re-read EVERY stripe (nslice x B*H fp32), so the
  // generic 9-slice policy made the fused path read 360 KB per WG and
  // lose to the separate reduce kernel (docs/KERNELS.md round-1 dead
  // end); 4 fatter slices keep the gemm parallel enough while the
  // head reads 160 KB
  const int nx = (H + 63) / 64, ny = (B + 63) / 64;
  int nslice = 1, kc = 0;
  if (nx * ny < 64 && K1 >= 256) {
    int want = 4;
    if (const char* e = getenv("TFA_HEAD_NSLICE")) want = atoi(e);
    if (want > K1 / 64) want = K1 / 64;
    if (want > 16) want = 16;
    if (want > 1) {
      kc = ((K1 + want - 1) / want + 31) / 32 * 32;
      nslice = (K1 + kc - 1) / kc;
    }
  }
  if (nslice > 1) {
    int* cnt_unused;
    float* ws = splitk_ws(x.device(), (long)B * H * nslice,
                          (long)nx * ny, &cnt_unused);
    launch_gemm_stripes((const bf16_t*)x.data_ptr(),
                        (const bf16_t*)w1.data_ptr(), ws, kc, nslice,
                        B, H, K1, K1, H, H,
                        vec_level(x.data_ptr(), K1),
                        vec_level(w1.data_ptr(), H), cur_stream());
    launch_mlp_fwd_head(ws, nslice, (const bf16_t*)b1.data_ptr(),
                        (const bf16_t*)w2.data_ptr(),
                        (const bf16_t*)b2.data_ptr(),
                        labels.data_ptr<long>(),
                        (bf16_t*)dlogits.data_ptr(),
                        (bf16_t*)dh.data_ptr(), loss.data_ptr<float>(),
                        dw2p, db2p, gf32, (float)scale, B, H, C,
                        cur_stream());
    return {loss, dh};
  }
  // no split-K at this shape: materialize h, run the normal head
  auto h = torch::empty({B, H}, x.options());
  launch_gemm((const bf16_t*)x.data_ptr(), (const bf16_t*)w1.data_ptr(),
              b1.data_ptr(), true, h.data_ptr(), false, nullptr, nullptr,
              nullptr, nullptr, 0, 1, B, H, K1, K1, H, H, false, false, 1,
              vec_level(x.data_ptr(), K1), vec_level(w1.data_ptr(), H),
              cur_stream());
  launch_mlp_head_fused((const bf16_t*)h.data_ptr(),
                        (const bf16_t*)w2.data_ptr(),
                        (const bf16_t*)b2.data_ptr(),
                        labels.data_ptr<long>(),
                        (bf16_t*)dlogits.data_ptr(), (bf16_t*)dh.data_ptr(),
                        loss.data_ptr<float>(), dw2p, db2p, gf32,
                        (float)scale, B, H, C, cur_stream());
  return {loss, dh};
}

torch::Tensor softmax_xent_bwd(torch::Tensor probs, torch::Tensor labels,
                               double scale) {
  TORCH_CHECK(probs.is_cuda() && probs.scalar_type() == torch::kBFloat16 &&
              probs.is_contiguous() && probs.dim() == 2, "probs bf16 [B,C]");
  int B = probs.size(0), C = probs.size(1);
  auto d = torch::empty_like(probs);
  launch_softmax_xent_bwd((const bf16_t*)probs.data_ptr(),
                          labels.data_ptr<long>(), (bf16_t*)d.data_ptr(),
                          (float)scale, B, C, cur_stream());
  return d;
}

torch::Tensor embedding_gather(torch::Tensor table, torch::Tensor ids) {
  TORCH_CHECK(table.is_cuda() && table.dim() == 2 && table.is_contiguous(),
              "table must be contiguous [V,D] on GPU");
  TORCH_CHECK(ids.scalar_type() == torch::kInt64 && ids.dim() == 1,
              "ids must be i64 [N]");
  long n = ids.numel(), V = table.size(0);
  int D = table.size(1);
  auto out = torch::empty({n, (long)D}, table.options());
  if (n == 0) return out;
  if (table.scalar_type() == torch::kBFloat16)
    launch_gather_bf16((const bf16_t*)table.data_ptr(), ids.data_ptr<long>(),
                       (bf16_t*)out.data_ptr(), n, D, V, cur_stream());
  else if (table.scalar_type() == torch::kFloat32)
    launch_gather_f32(table.data_ptr<float>(), ids.data_ptr<long>(),
                      out.data_ptr<float>(), n, D, V, cur_stream());
  else
    TORCH_CHECK(false, "table must be bf16 or fp32");
  return out;
}

void embedding_scatter_add(torch::Tensor table, torch::Tensor ids,
                           torch::Tensor rows) {
  TORCH_CHECK(table.is_cuda() && table.scalar_type() == torch::kFloat32 &&
              table.is_contiguous(), "scatter-add table must be fp32 [V,D]");
  TORCH_CHECK(ids.scalar_type() == torch::kInt64, "ids must be i64");
  long n = ids.numel(), V = table.size(0);
  int D = table.size(1);
  if (n == 0) return;
  TORCH_CHECK(rows.is_contiguous() && rows.numel() == n * D,
              "rows shape mismatch");
  if (rows.scalar_type() == torch::kBFloat16)
    launch_scatter_add_bf16(table.data_ptr<float>(), ids.data_ptr<long>(),
                            (const bf16_t*)rows.data_ptr(), n, D, V,
                            cur_stream());
  else
    launch_scatter_add_f32(table.data_ptr<float>(), ids.data_ptr<long>(),
                           rows.data_ptr<float>(), n, D, V, cur_stream());
}

torch::Tensor relu_bwd(torch::Tensor dy, torch::Tensor act) {
  TORCH_CHECK(dy.is_cuda() && dy.scalar_type() == torch::kBFloat16 &&
              dy.is_contiguous() && act.is_contiguous() &&
              act.numel() == dy.numel(), "relu_bwd: bf16 contiguous");
  auto dx = torch::empty_like(dy);
  launch_relu_bwd((const bf16_t*)dy.data_ptr(), (const bf16_t*)act.data_ptr(),
                  (bf16_t*)dx.data_ptr(), dy.numel(), cur_stream());
  return dx;
}

torch::Tensor colsum(torch::Tensor x, torch::Tensor out) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16 &&
              x.is_contiguous() && x.dim() == 2, "colsum: bf16 [M,N]");
  int M = x.size(0), N = x.size(1);
  if (out.numel() == 0)
    out = torch::empty({N}, x.options().dtype(torch::kFloat32));
  else
    TORCH_CHECK(out.scalar_type() == torch::kFloat32 && out.numel() == N &&
                out.is_contiguous(), "colsum out must be fp32 [N]");
  launch_colsum((const bf16_t*)x.data_ptr(), out.data_ptr<float>(), M, N,
                cur_stream());
  return out;
}

// ------------------------------------------------------------------ conv

static bool is_cl(const torch::Tensor& t) {
  return t.is_contiguous(torch::MemoryFormat::ChannelsLast);
}

// x: logical NCHW in channels-last memory; wm: [K,R,S,C] contiguous
// (tap-major weight copy built by the Python wrapper). Returns y in
// channels-last memory.
torch::Tensor conv2d_fwd(torch::Tensor x, torch::Tensor wm,
                         torch::Tensor bias, long stride_h, long stride_w,
                         long pad_h, long pad_w, bool relu) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 && is_cl(x) &&
              x.scalar_type() == torch::kBFloat16,
              "x must be bf16 NCHW in channels-last memory");
  TORCH_CHECK(wm.is_cuda() && wm.dim() == 4 && wm.is_contiguous() &&
              wm.scalar_type() == torch::kBFloat16 &&
              wm.size(3) == x.size(1), "wm must be bf16 [K,R,S,C]");
  const long K = wm.size(0), R = wm.size(1), S = wm.size(2);
  const float* bias_p = nullptr;
  if (bias.numel() > 0) {
    TORCH_CHECK(bias.scalar_type() == torch::kFloat32 && bias.numel() == K,
                "bias must be fp32 [K]");
    bias_p = bias.data_ptr<float>();
  }
  const int Ho = (int)((x.size(2) + 2 * pad_h - R) / stride_h + 1);
  const int Wo = (int)((x.size(3) + 2 * pad_w - S) / stride_w + 1);
  auto y = torch::empty({x.size(0), K, Ho, Wo},
                        x.options().memory_format(
                            torch::MemoryFormat::ChannelsLast));
  float* ws = nullptr;
  const int z = conv_fwd_slices(x.size(0), K, Ho, Wo, x.size(1), R, S);
  if (z > 1) {
    int* cnt_unused;
    ws = splitk_ws(x.device(), (long)x.size(0) * Ho * Wo * K * z, 1,
                   &cnt_unused);
  }
  launch_conv_fwd((const bf16_t*)x.data_ptr(), (const bf16_t*)wm.data_ptr(),
                  bias_p, (bf16_t*)y.data_ptr(), ws, x.size(0), x.size(1),
                  x.size(2), x.size(3), K, R, S, Ho, Wo, stride_h, stride_w,
                  pad_h, pad_w, relu, cur_stream());
  return y;
}

// dy channels-last; wt: [C,R,S,K] contiguous (W^T copy). Returns dx
// channels-last.
static bool cl_narrow(const torch::Tensor& t, long* ldy);

// dy may be a channel-narrow view of the block concat-grad buffer
// (_JoinViews backward): the kernels read it strided (ConvShape.LDY),
// so the terminal convs of every Inception block skip a per-layer
// contiguous() copy of their dy slice.
torch::Tensor conv2d_bwd_data(torch::Tensor dy, torch::Tensor wt,
                              long H, long W, long stride_h, long stride_w,
                              long pad_h, long pad_w) {
  long ldy = 0;
  TORCH_CHECK(dy.is_cuda() && cl_narrow(dy, &ldy) &&
              dy.scalar_type() == torch::kBFloat16,
              "dy must be bf16 channels-last (or channel-narrow)");
  // weight in its NATIVE [K,R,S,C] layout — the kernel's transposing
  // stager reads it directly (no host-side W^T permute per call)
  TORCH_CHECK(wt.is_contiguous() && wt.dim() == 4 &&
              wt.size(0) == dy.size(1), "w must be [K,R,S,C]");
  const long C = wt.size(3), R = wt.size(1), S = wt.size(2);
  auto dx = torch::empty({dy.size(0), C, H, W},
                         dy.options().memory_format(
                             torch::MemoryFormat::ChannelsLast));
  float* ws = nullptr;
  const int z = conv_bwdd_slices(dy.size(0), H, W, C, dy.size(1), R, S);
  if (z > 1) {
    int* cnt_unused;
    ws = splitk_ws(dy.device(), (long)dy.size(0) * H * W * C * z, 1,
                   &cnt_unused);
  }
  launch_conv_bwd_data((const bf16_t*)dy.data_ptr(), ldy,
                       (const bf16_t*)wt.data_ptr(), (bf16_t*)dx.data_ptr(),
                       ws, dy.size(0), C, H, W, dy.size(1), R, S,
                       dy.size(2), dy.size(3), stride_h, stride_w,
                       pad_h, pad_w, cur_stream());
  return dx;
}

// Returns dW in tap-major [K,R,S,C] fp32 (wrapper permutes to KCRS).
torch::Tensor conv2d_bwd_weight(torch::Tensor dy, torch::Tensor x,
                                long R, long S, long stride_h, long stride_w,
                                long pad_h, long pad_w) {
  long ldy = 0;
  TORCH_CHECK(dy.is_cuda() && x.is_cuda() && cl_narrow(dy, &ldy) && is_cl(x),
              "dy must be channels-last (or channel-narrow), x channels-last");
  // zero-init only when the reduction is z-sliced (atomic accumulate)
  auto opts = x.options().dtype(torch::kFloat32);
  const int z = conv_bwdw_slices(x.size(0), x.size(1), dy.size(1), R, S,
                                 dy.size(2), dy.size(3));
  auto dw = z > 1 ? torch::zeros({dy.size(1), R, S, x.size(1)}, opts)
                  : torch::empty({dy.size(1), R, S, x.size(1)}, opts);
  launch_conv_bwd_weight((const bf16_t*)dy.data_ptr(), ldy,
                         (const bf16_t*)x.data_ptr(), dw.data_ptr<float>(),
                         x.size(0), x.size(1), x.size(2), x.size(3),
                         dy.size(1), R, S, dy.size(2), dy.size(3), stride_h,
                         stride_w, pad_h, pad_w, cur_stream());
  return dw;
}

// Accumulates dW into a caller-owned PRE-ZEROED fp32 [K,R,S,C] buffer
// (the per-model grad arena: one bulk zero per backward replaces ~90
// per-layer fills, and the fp32 result feeds the trainer's batched
// bf16 gather copy — no per-layer cast kernels either).
void conv2d_bwd_weight_out(torch::Tensor dy, torch::Tensor x,
                           long R, long S, long stride_h, long stride_w,
                           long pad_h, long pad_w, torch::Tensor dw) {
  long ldy = 0;
  TORCH_CHECK(dy.is_cuda() && x.is_cuda() && cl_narrow(dy, &ldy) && is_cl(x),
              "dy must be channels-last (or channel-narrow), x channels-last");
  TORCH_CHECK(dw.is_cuda() && dw.is_contiguous() &&
              dw.scalar_type() == torch::kFloat32 && dw.dim() == 4 &&
              dw.size(0) == dy.size(1) && dw.size(1) == R &&
              dw.size(2) == S && dw.size(3) == x.size(1),
              "dw must be contiguous fp32 [K,R,S,C]");
  launch_conv_bwd_weight((const bf16_t*)dy.data_ptr(), ldy,
                         (const bf16_t*)x.data_ptr(), dw.data_ptr<float>(),
                         x.size(0), x.size(1), x.size(2), x.size(3),
                         dy.size(1), R, S, dy.size(2), dy.size(3), stride_h,
                         stride_w, pad_h, pad_w, cur_stream());
}

// -------------------------------------------------------------------- bn

// dy may be a channel-narrow view of a wider channels-last tensor (the
// backward of torch.cat): detected by its strides and read in place —
// no contiguous() copy.
static bool cl_narrow(const torch::Tensor& t, long* ldy) {
  if (t.dim() != 4) return false;
  const auto s = t.strides();
  const long C = t.size(1), H = t.size(2), W = t.size(3);
  if (s[1] != 1 || s[3] < C || s[2] != W * s[3] || s[0] != H * s[2])
    return false;
  *ldy = s[3];
  return true;
}

std::vector<torch::Tensor> bn_fwd(torch::Tensor x, torch::Tensor g,
                                  torch::Tensor b, double eps, bool relu,
                                  torch::Tensor out) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 && is_cl(x) &&
              x.scalar_type() == torch::kBFloat16,
              "x must be bf16 channels-last");
  const int C = x.size(1);
  const long P = x.numel() / C;
  TORCH_CHECK(C <= bn_max_channels(), "bn: C > LDS staging bound");
  TORCH_CHECK(g.numel() == C && b.numel() == C &&
              g.scalar_type() == torch::kBFloat16 &&
              b.scalar_type() == torch::kBFloat16, "g/b must be bf16 [C]");
  const int Z = bn_stats_slices(P, C);
  auto opts = x.options().dtype(torch::kFloat32);
  // out (optional): a channel-narrow channels-last view — the apply
  // writes straight into the caller's concat buffer (strided store)
  long ldo = C;
  torch::Tensor y;
  if (out.numel() > 0) {
    TORCH_CHECK(cl_narrow(out, &ldo) && out.sizes() == x.sizes() &&
                out.scalar_type() == torch::kBFloat16,
                "bn_fwd: out must be a bf16 channels-last (or channel-"
                "narrow) tensor with x's shape");
    y = out;
  } else {
    y = torch::empty_like(x);
  }
  auto mean = torch::empty({C}, opts);
  auto invstd = torch::empty({C}, opts);
  auto part = torch::empty({(long)Z * C * 2}, opts);
  launch_bn_fwd((const bf16_t*)x.data_ptr(), (const bf16_t*)g.data_ptr(),
                (const bf16_t*)b.data_ptr(), (bf16_t*)y.data_ptr(), ldo,
                mean.data_ptr<float>(), invstd.data_ptr<float>(),
                part.data_ptr<float>(), P, C, Z, (float)eps, relu,
                cur_stream());
  return {y, mean, invstd};
}

std::vector<torch::Tensor> bn_bwd(torch::Tensor x, torch::Tensor dy,
                                  torch::Tensor g, torch::Tensor b,
                                  torch::Tensor mean, torch::Tensor invstd,
                                  bool relu) {
  long ldy = 0;
  TORCH_CHECK(is_cl(x), "bn_bwd: channels-last x required");
  TORCH_CHECK(cl_narrow(dy, &ldy), "bn_bwd: dy must be channels-last or a "
              "channel-narrow view of a channels-last tensor");
  const int C = x.size(1);
  const long P = x.numel() / C;
  const int Z = bn_stats_slices(P, C);
  auto opts = x.options().dtype(torch::kFloat32);
  auto dx = torch::empty_like(x);
  auto dgamma = torch::empty({C}, x.options());
  auto dbeta = torch::empty({C}, x.options());
  auto part = torch::empty({(long)Z * C * 2}, opts);
  auto s1n = torch::empty({C}, opts);
  auto s2n = torch::empty({C}, opts);
  launch_bn_bwd((const bf16_t*)x.data_ptr(), (const bf16_t*)dy.data_ptr(),
                ldy, (const bf16_t*)g.data_ptr(), (const bf16_t*)b.data_ptr(),
                mean.data_ptr<float>(), invstd.data_ptr<float>(),
                (bf16_t*)dx.data_ptr(), (bf16_t*)dgamma.data_ptr(),
                (bf16_t*)dbeta.data_ptr(), part.data_ptr<float>(),
                s1n.data_ptr<float>(), s2n.data_ptr<float>(), P, C, Z,
                relu, cur_stream());
  return {dx, dgamma, dbeta};
}


// grouped BN: ONE stats/finalize/apply launch triple for all terminal
// BNs of an Inception block (per-channel math — identical results to
// separate per-branch BNs, ~6x fewer launches per block)
std::vector<torch::Tensor> bn_group_fwd(std::vector<torch::Tensor> xs,
                                        std::vector<torch::Tensor> gs,
                                        std::vector<torch::Tensor> bs,
                                        torch::Tensor out, double eps,
                                        bool relu) {
  const int n = (int)xs.size();
  TORCH_CHECK(n >= 1 && n <= 8 && (int)gs.size() == n &&
              (int)bs.size() == n, "bn_group: 1..8 branches");
  const bf16_t* xp[8];
  const bf16_t* gp[8];
  const bf16_t* bp[8];
  int Cs[8];
  int ctot = 0;
  const long P = xs[0].numel() / xs[0].size(1);
  for (int i = 0; i < n; ++i) {
    auto& x = xs[i];
    TORCH_CHECK(x.is_cuda() && x.dim() == 4 && is_cl(x) &&
                x.scalar_type() == torch::kBFloat16 &&
                x.numel() / x.size(1) == P && (x.size(1) & 7) == 0 &&
                x.size(1) <= 512,
                "bn_group_fwd: branch inputs must be bf16 channels-last "
                "with equal spatial size, C%8==0, C<=512");
    Cs[i] = (int)x.size(1);
    TORCH_CHECK(gs[i].numel() == Cs[i] && bs[i].numel() == Cs[i] &&
                gs[i].scalar_type() == torch::kBFloat16 &&
                bs[i].scalar_type() == torch::kBFloat16, "bad g/b");
    xp[i] = (const bf16_t*)x.data_ptr();
    gp[i] = (const bf16_t*)gs[i].data_ptr();
    bp[i] = (const bf16_t*)bs[i].data_ptr();
    ctot += Cs[i];
  }
  long ldo = ctot;
  TORCH_CHECK(cl_narrow(out, &ldo) && out.size(1) == ctot &&
              out.numel() / out.size(1) == P &&
              out.scalar_type() == torch::kBFloat16,
              "bn_group_fwd: out must be a bf16 channels-last (or "
              "channel-narrow) [N,Ctot,H,W] view");
  const int Z = bn_group_slices(P, Cs, n);
  auto opts = xs[0].options().dtype(torch::kFloat32);
  auto mean = torch::empty({ctot}, opts);
  auto invstd = torch::empty({ctot}, opts);
  auto part = torch::empty({(long)Z * ctot * 2}, opts);
  bf16_t* youts[8];
  long ylds[8];
  int coff = 0;
  for (int i = 0; i < n; ++i) {
    youts[i] = (bf16_t*)out.data_ptr() + coff;
    ylds[i] = ldo;
    coff += Cs[i];
  }
  launch_bn_group_fwd(xp, gp, bp, Cs, n, youts, ylds,
                      mean.data_ptr<float>(), invstd.data_ptr<float>(),
                      part.data_ptr<float>(), P, Z, (float)eps, relu,
                      cur_stream());
  return {mean, invstd};
}

// per-branch DENSE outputs (parallel inner-stage BNs whose results
// feed different consumers): same grouped kernels, one launch triple
std::vector<torch::Tensor> bn_group_fwd_multi(std::vector<torch::Tensor> xs,
                                              std::vector<torch::Tensor> gs,
                                              std::vector<torch::Tensor> bs,
                                              double eps, bool relu) {
  const int n = (int)xs.size();
  TORCH_CHECK(n >= 1 && n <= 8, "bn_group: 1..8 branches");
  const bf16_t* xp[8];
  const bf16_t* gp[8];
  const bf16_t* bp[8];
  bf16_t* youts[8];
  long ylds[8];
  int Cs[8];
  int ctot = 0;
  const long P = xs[0].numel() / xs[0].size(1);
  std::vector<torch::Tensor> outs;
  for (int i = 0; i < n; ++i) {
    auto& x = xs[i];
    TORCH_CHECK(x.is_cuda() && x.dim() == 4 && is_cl(x) &&
                x.scalar_type() == torch::kBFloat16 &&
                x.numel() / x.size(1) == P && (x.size(1) & 7) == 0 &&
                x.size(1) <= 512, "bn_group_fwd_multi: bad branch input");
    Cs[i] = (int)x.size(1);
    xp[i] = (const bf16_t*)x.data_ptr();
    gp[i] = (const bf16_t*)gs[i].data_ptr();
    bp[i] = (const bf16_t*)bs[i].data_ptr();
    auto y = torch::empty_like(x);
    youts[i] = (bf16_t*)y.data_ptr();
    ylds[i] = Cs[i];
    outs.push_back(y);
    ctot += Cs[i];
  }
  const int Z = bn_group_slices(P, Cs, n);
  auto opts = xs[0].options().dtype(torch::kFloat32);
  auto mean = torch::empty({ctot}, opts);
  auto invstd = torch::empty({ctot}, opts);
  auto part = torch::empty({(long)Z * ctot * 2}, opts);
  launch_bn_group_fwd(xp, gp, bp, Cs, n, youts, ylds,
                      mean.data_ptr<float>(), invstd.data_ptr<float>(),
                      part.data_ptr<float>(), P, Z, (float)eps, relu,
                      cur_stream());
  outs.push_back(mean);
  outs.push_back(invstd);
  return outs;
}

std::vector<torch::Tensor> bn_group_bwd(std::vector<torch::Tensor> xs,
                                        torch::Tensor dy,
                                        std::vector<torch::Tensor> gs,
                                        std::vector<torch::Tensor> bs,
                                        torch::Tensor mean,
                                        torch::Tensor invstd, bool relu) {
  const int n = (int)xs.size();
  TORCH_CHECK(n >= 1 && n <= 8, "bn_group: 1..8 branches");
  const bf16_t* xp[8];
  const bf16_t* gp[8];
  const bf16_t* bp[8];
  bf16_t* dxp[8];
  int Cs[8];
  int ctot = 0;
  const long P = xs[0].numel() / xs[0].size(1);
  std::vector<torch::Tensor> outs;
  for (int i = 0; i < n; ++i) {
    Cs[i] = (int)xs[i].size(1);
    xp[i] = (const bf16_t*)xs[i].data_ptr();
    gp[i] = (const bf16_t*)gs[i].data_ptr();
    bp[i] = (const bf16_t*)bs[i].data_ptr();
    auto dx = torch::empty_like(xs[i]);
    dxp[i] = (bf16_t*)dx.data_ptr();
    outs.push_back(dx);
    ctot += Cs[i];
  }
  long ldy = ctot;
  TORCH_CHECK(dy.is_cuda() && cl_narrow(dy, &ldy) && dy.size(1) == ctot,
              "bn_group_bwd: dy must be channels-last or channel-narrow");
  const bf16_t* dys[8];
  long dylds[8];
  int coff = 0;
  for (int i = 0; i < n; ++i) {
    dys[i] = (const bf16_t*)dy.data_ptr() + coff;
    dylds[i] = ldy;
    coff += Cs[i];
  }
  const int Z = bn_group_slices(P, Cs, n);
  auto opts = xs[0].options().dtype(torch::kFloat32);
  auto dgamma = torch::empty({ctot}, xs[0].options());
  auto dbeta = torch::empty({ctot}, xs[0].options());
  auto s1n = torch::empty({ctot}, opts);
  auto s2n = torch::empty({ctot}, opts);
  auto part = torch::empty({(long)Z * ctot * 2}, opts);
  launch_bn_group_bwd(xp, dys, dylds, gp, bp, dxp,
                      Cs, n, mean.data_ptr<float>(),
                      invstd.data_ptr<float>(), (bf16_t*)dgamma.data_ptr(),
                      (bf16_t*)dbeta.data_ptr(), s1n.data_ptr<float>(),
                      s2n.data_ptr<float>(), part.data_ptr<float>(), P, Z,
                      relu, cur_stream());
  outs.push_back(dgamma);
  outs.push_back(dbeta);
  return outs;
}

std::vector<torch::Tensor> bn_group_bwd_multi(std::vector<torch::Tensor> xs,
                                              std::vector<torch::Tensor> dys_in,
                                              std::vector<torch::Tensor> gs,
                                              std::vector<torch::Tensor> bs,
                                              torch::Tensor mean,
                                              torch::Tensor invstd,
                                              bool relu) {
  const int n = (int)xs.size();
  TORCH_CHECK(n >= 1 && n <= 8 && (int)dys_in.size() == n,
              "bn_group: 1..8 branches");
  const bf16_t* xp[8];
  const bf16_t* gp[8];
  const bf16_t* bp[8];
  bf16_t* dxp[8];
  const bf16_t* dys[8];
  long dylds[8];
  int Cs[8];
  int ctot = 0;
  const long P = xs[0].numel() / xs[0].size(1);
  std::vector<torch::Tensor> outs;
  for (int i = 0; i < n; ++i) {
    Cs[i] = (int)xs[i].size(1);
    xp[i] = (const bf16_t*)xs[i].data_ptr();
    gp[i] = (const bf16_t*)gs[i].data_ptr();
    bp[i] = (const bf16_t*)bs[i].data_ptr();
    long ldy = Cs[i];
    TORCH_CHECK(cl_narrow(dys_in[i], &ldy) && dys_in[i].size(1) == Cs[i],
                "bn_group_bwd_multi: bad dy");
    dys[i] = (const bf16_t*)dys_in[i].data_ptr();
    dylds[i] = ldy;
    auto dx = torch::empty_like(xs[i]);
    dxp[i] = (bf16_t*)dx.data_ptr();
    outs.push_back(dx);
    ctot += Cs[i];
  }
  const int Z = bn_group_slices(P, Cs, n);
  auto opts = xs[0].options().dtype(torch::kFloat32);
  auto dgamma = torch::empty({ctot}, xs[0].options());
  auto dbeta = torch::empty({ctot}, xs[0].options());
  auto s1n = torch::empty({ctot}, opts);
  auto s2n = torch::empty({ctot}, opts);
  auto part = torch::empty({(long)Z * ctot * 2}, opts);
  launch_bn_group_bwd(xp, dys, dylds, gp, bp, dxp,
                      Cs, n, mean.data_ptr<float>(),
                      invstd.data_ptr<float>(), (bf16_t*)dgamma.data_ptr(),
                      (bf16_t*)dbeta.data_ptr(), s1n.data_ptr<float>(),
                      s2n.data_ptr<float>(), part.data_ptr<float>(), P, Z,
                      relu, cur_stream());
  outs.push_back(dgamma);
  outs.push_back(dbeta);
  return outs;
}

torch::Tensor avg_pool3x3(torch::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 && is_cl(x) &&
              x.scalar_type() == torch::kBFloat16,
              "avg_pool3x3: bf16 channels-last NCHW");
  auto y = torch::empty_like(x);
  launch_avg3x3((const bf16_t*)x.data_ptr(), (bf16_t*)y.data_ptr(),
                x.size(0), x.size(2), x.size(3), x.size(1), cur_stream());
  return y;
}

std::vector<torch::Tensor> maxpool3x3s2_fwd(torch::Tensor x,
                                            torch::Tensor out) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 && is_cl(x) &&
              x.scalar_type() == torch::kBFloat16,
              "maxpool3x3s2: bf16 channels-last NCHW");
  const int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  const int Ho = (H - 3) / 2 + 1, Wo = (W - 3) / 2 + 1;
  long ldo = C;
  torch::Tensor y;
  if (out.numel() > 0) {
    // channel-narrow view of the caller's concat buffer: the kernel
    // stores strided and the aten cat copy disappears (like bn_fwd)
    TORCH_CHECK(cl_narrow(out, &ldo) && out.size(0) == N &&
                out.size(1) == C && out.size(2) == Ho && out.size(3) == Wo &&
                out.scalar_type() == torch::kBFloat16,
                "maxpool3x3s2_fwd: out must be a bf16 channels-last (or "
                "channel-narrow) [N,C,Ho,Wo] tensor");
    y = out;
  } else {
    y = torch::empty({N, C, Ho, Wo},
                     x.options().memory_format(
                         torch::MemoryFormat::ChannelsLast));
  }
  auto idx = torch::empty({N, C, Ho, Wo},
                          x.options().dtype(torch::kUInt8).memory_format(
                              torch::MemoryFormat::ChannelsLast));
  launch_maxpool3x3s2_fwd((const bf16_t*)x.data_ptr(), (bf16_t*)y.data_ptr(),
                          ldo, idx.data_ptr<unsigned char>(), N, H, W, C, Ho,
                          Wo, cur_stream());
  return {y, idx};
}

torch::Tensor maxpool3x3s2_bwd(torch::Tensor dy, torch::Tensor idx,
                               long H, long W) {
  long ldy = 0;
  TORCH_CHECK(dy.is_cuda() && cl_narrow(dy, &ldy) && is_cl(idx),
              "maxpool bwd: dy must be channels-last or a channel-narrow "
              "view of one");
  const int N = dy.size(0), C = dy.size(1);
  auto dx = torch::empty({N, C, H, W},
                         dy.options().memory_format(
                             torch::MemoryFormat::ChannelsLast));
  launch_maxpool3x3s2_bwd((const bf16_t*)dy.data_ptr(), ldy,
                          idx.data_ptr<unsigned char>(),
                          (bf16_t*)dx.data_ptr(), N, H, W, C, dy.size(2),
                          dy.size(3), cur_stream());
  return dx;
}

}  // namespace

// mnist single-GPU fused tail: dW1 = x^T @ dh with the SGD apply of
// ALL FOUR params in the epilogue (W1/b1 from this GEMM's fp32
// accumulators; W2/b2 by re-reading the small classifier grads the
// head kernel just wrote). Replaces [dW1 GEMM, flat sgd_kernel] with
// ONE launch; valid for plain SGD, grad_scale 1 (world==1 colocated).
// Reference parity: the one-train-op apply of mnist_replica.py:146-147
// collapsed into the producing kernels.
void mlp_tail_sgd(torch::Tensor x, torch::Tensor dh,
                  torch::Tensor w1m, torch::Tensor w1s,
                  torch::Tensor b1m, torch::Tensor b1s,
                  torch::Tensor gw2, torch::Tensor w2m, torch::Tensor w2s,
                  torch::Tensor gb2, torch::Tensor b2m, torch::Tensor b2s,
                  double lr) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 && x.is_contiguous() &&
              x.scalar_type() == torch::kBFloat16, "x must be bf16 [B,M]");
  TORCH_CHECK(dh.is_cuda() && dh.dim() == 2 && dh.is_contiguous() &&
              dh.scalar_type() == torch::kBFloat16 &&
              dh.size(0) == x.size(0), "dh must be bf16 [B,N]");
  const int B = (int)x.size(0), M = (int)x.size(1), N = (int)dh.size(1);
  for (auto* t : {&w1m, &b1m, &w2m, &b2m})
    TORCH_CHECK(t->is_cuda() && t->is_contiguous() &&
                t->scalar_type() == torch::kFloat, "masters must be fp32");
  for (auto* t : {&w1s, &b1s, &gw2, &w2s, &gb2, &b2s})
    TORCH_CHECK(t->is_cuda() && t->is_contiguous() &&
                t->scalar_type() == torch::kBFloat16,
                "shadows/grads must be bf16");
  TORCH_CHECK(w1m.numel() == (long)M * N && b1m.numel() == N,
              "W1/b1 shape mismatch");
  TORCH_CHECK(gw2.numel() == w2m.numel() && gb2.numel() == b2m.numel(),
              "W2/b2 grad shape mismatch");
  const long stiles = (long)((M + 31) / 32) * ((N + 31) / 32);
  TORCH_CHECK(stiles >= 2 && stiles <= 512 && B <= 256,
              "shape outside the small-GEMM regime");
  SmallSgdArgs sga;
  sga.pmw = (float*)w1m.data_ptr();
  sga.psw = w1s.data_ptr();
  sga.pmb = (float*)b1m.data_ptr();
  sga.psb = b1s.data_ptr();
  sga.g2w = gw2.data_ptr();
  sga.pm2w = (float*)w2m.data_ptr();
  sga.ps2w = w2s.data_ptr();
  sga.g2b = gb2.data_ptr();
  sga.pm2b = (float*)b2m.data_ptr();
  sga.ps2b = b2s.data_ptr();
  sga.lr = (float)lr;
  sga.n2w = (int)w2m.numel();
  sga.n2b = (int)b2m.numel();
  launch_gemm_small((const bf16_t*)x.data_ptr(),
                    (const bf16_t*)dh.data_ptr(), nullptr, false,
                    nullptr, false, 0, nullptr, false,
                    M, N, B, (int)x.size(1), (int)dh.size(1), N,
                    /*ta=*/true, &sga, cur_stream());
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("fused_sgd", &fused_sgd, "fused SGD apply + bf16 shadow",
        py::arg("param"), py::arg("grad"), py::arg("momentum_buf"),
        py::arg("bf16_out"), py::arg("lr"), py::arg("momentum") = 0.0,
        py::arg("weight_decay") = 0.0, py::arg("grad_scale") = 1.0,
        py::arg("neg_decay") = 0.0);
  m.def("fused_adam", &fused_adam, "fused Adam apply + bf16 shadow",
        py::arg("param"), py::arg("grad"), py::arg("m"), py::arg("v"),
        py::arg("bf16_out"), py::arg("step"), py::arg("lr"),
        py::arg("beta1") = 0.9, py::arg("beta2") = 0.999,
        py::arg("eps") = 1e-8, py::arg("weight_decay") = 0.0,
        py::arg("grad_scale") = 1.0);
  m.def("fused_adagrad", &fused_adagrad, "fused Adagrad apply + bf16 shadow",
        py::arg("param"), py::arg("grad"), py::arg("accum"),
        py::arg("bf16_out"), py::arg("lr"), py::arg("eps") = 1e-10,
        py::arg("weight_decay") = 0.0, py::arg("grad_scale") = 1.0);
  m.def("gemm_bias_act", &gemm_bias_act, "bf16 MFMA GEMM + bias + act");
  m.def("gemm_bias_act_out", &gemm_bias_act_out,
        "bf16 MFMA GEMM + fused epilogue (bias/act/relu_bwd/colsum) into "
        "out (bf16 or fp32), split-K for deep skinny shapes");
  m.def("softmax_xent_fused", &softmax_xent_fused);
  m.def("mlp_head_fused", &mlp_head_fused);
  m.def("head_debug", [](torch::Tensor t) {
    set_head_debug(t.numel() ? t.data_ptr() : nullptr);
  });
  m.def("mlp_fwd_head_fused", &mlp_fwd_head_fused);
  m.def("softmax_xent_fwd", &softmax_xent_fwd);
  m.def("softmax_xent_bwd", &softmax_xent_bwd);
  m.def("conv2d_fwd", &conv2d_fwd, "implicit-GEMM conv fwd (MFMA, bf16)");
  m.def("bn_fwd", &bn_fwd, "fused train-mode batch-norm (+relu) fwd");
  m.def("avg_pool3x3", &avg_pool3x3, "3x3 s1 p1 avg pool, channels-last");
  m.def("maxpool3x3s2_fwd", &maxpool3x3s2_fwd);
  m.def("bn_group_fwd", &bn_group_fwd);
  m.def("bn_group_bwd", &bn_group_bwd);
  m.def("bn_group_fwd_multi", &bn_group_fwd_multi);
  m.def("gemm_sgd", &gemm_sgd);
  m.def("mlp_tail_sgd", &mlp_tail_sgd);
  m.def("gemm_sgd_pair", &gemm_sgd_pair);
  m.def("add_n", &add_n);
  m.def("bn_group_bwd_multi", &bn_group_bwd_multi);
  m.def("maxpool3x3s2_bwd", &maxpool3x3s2_bwd);
  m.def("bn_bwd", &bn_bwd, "fused batch-norm (+relu mask) bwd");
  m.def("conv2d_bwd_data", &conv2d_bwd_data);
  m.def("conv2d_bwd_weight", &conv2d_bwd_weight);
  m.def("conv2d_bwd_weight_out", &conv2d_bwd_weight_out);
  m.def("embedding_gather", &embedding_gather);
  m.def("embedding_scatter_add", &embedding_scatter_add);
  m.def("relu_bwd", &relu_bwd);
  m.def("colsum", &colsum, py::arg("x"), py::arg("out"));
}
