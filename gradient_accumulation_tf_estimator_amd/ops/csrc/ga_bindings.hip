// PyTorch bindings for the MI355X gradient-accumulation kernels.
// Native HIP path only (no CUDA shim, no hipify): launches on the current
// c10 HIP stream so torch.cuda.graphs capture works.

#include <cstdlib>
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

#define GA_THREADS 256
#define GA_MAX_BLOCKS 2048

extern "C" __global__ void k_accum_f32(float4*, float4*, long long);
extern "C" __global__ void k_accum_bf16(float4*, ushort4*, long long);
extern "C" __global__ void k_sqnorm(const float4*, long long, float*);
extern "C" __global__ void k_apply_f32(float4*, float4*, float4*, float4*,
                                       const float*, const float*,
                                       long long, long long,
                                       float, float, float, float, float, float);
extern "C" __global__ void k_apply_bf16(float4*, float4*, float4*, float4*, ushort4*,
                                        const float*, const float*,
                                        long long, long long,
                                        float, float, float, float, float, float);
extern "C" __global__ void k_apply_f32_g2(float4*, float4*, float4*, float4*,
                                          const float*, const float*,
                                          long long, long long,
                                          float, float, float, float, float, float);
extern "C" __global__ void k_apply_bf16_g2(float4*, float4*, float4*, float4*, ushort4*,
                                           const float*, const float*,
                                           long long, long long,
                                           float, float, float, float, float, float);
// The 2-group-ILP apply is the DEFAULT: the 4-group variant measured
// 19.6-19.8k vs 20.8-21.1k samples/s interleaved on one box (the extra
// in-flight registers cost occupancy). GA_APPLY_G4=1 re-measures it.
static inline bool apply_g2() {
  static const bool on = [] {
    const char* v = getenv("GA_APPLY_G4");
    return !(v && atoi(v) != 0);
  }();
  return on;
}
#define GA_DECL_LN(EPL)                                                                  \
  extern "C" __global__ void k_addln_fwd_##EPL(                                          \
      const unsigned short*, const unsigned short*, const unsigned short*,               \
      const unsigned short*, const unsigned short*, unsigned short*, unsigned short*,    \
      float*, float*, int, int, float);                                                  \
  extern "C" __global__ void k_addln_bwd_##EPL(                                          \
      const unsigned short*, const unsigned short*, const unsigned short*, const float*, \
      const float*, unsigned short*, float*, int, int);
GA_DECL_LN(4) GA_DECL_LN(8) GA_DECL_LN(12) GA_DECL_LN(16)
GA_DECL_LN(s2_4) GA_DECL_LN(s2_8)
#define GA_DECL_GELU(EPL)                                                 \
  extern "C" __global__ void k_biasgelu_bwd_##EPL(                        \
      const unsigned short*, const unsigned short*, const unsigned short*, \
      unsigned short*, float*, int, int);
GA_DECL_GELU(8) GA_DECL_GELU(12) GA_DECL_GELU(16) GA_DECL_GELU(24) GA_DECL_GELU(32) GA_DECL_GELU(48) GA_DECL_GELU(64)
extern "C" __global__ void k_biasgelu_fwd(const unsigned short*, const unsigned short*,
                                          unsigned short*, long long, int);
extern "C" __global__ void k_biasgelu_bwd_ew(const unsigned short*, const unsigned short*,
                                             const unsigned short*, unsigned short*,
                                             long long, int);
extern "C" __global__ void k_lin_fwd_bf16(const unsigned short*, const unsigned short*,
                                          const unsigned short*, unsigned short*,
                                          int, int, int);
extern "C" __global__ void k_lin_fwd_f32(const unsigned short*, const unsigned short*,
                                         float*, int, int, int, int);
extern "C" __global__ void k_lin_dgrad_bf16(const unsigned short*, const unsigned short*,
                                            unsigned short*, int, int, int);
extern "C" __global__ void k_lin_dgrad_f32(const unsigned short*, const unsigned short*,
                                           float*, int, int, int, int);
extern "C" __global__ void k_splitk_combine(const float*, int, long long, int,
                                            const unsigned short*, unsigned short*);
extern "C" __global__ void k_colreduce_acc(const float*, int, int, float*, int,
                                           float*, int, float*);
#define CRB_MAX_G 16
struct CrbArgs {
  unsigned long long part[CRB_MAX_G];
  unsigned long long d0[CRB_MAX_G];
  unsigned long long d1[CRB_MAX_G];
  unsigned long long d2[CRB_MAX_G];
  int nb[CRB_MAX_G];
  int c[CRB_MAX_G];
  int n0[CRB_MAX_G];
  int n1[CRB_MAX_G];
  int G;
};
extern "C" __global__ void k_colreduce_batch(CrbArgs);
extern "C" __global__ void k_embgrad_acc(const unsigned short*, const long long*,
                                         float*, long long, int);
extern "C" __global__ void k_emb3_fwd(const long long*, const long long*,
                                      const unsigned short*, const unsigned short*,
                                      const unsigned short*, unsigned short*,
                                      int, int);
#define GA_DECL_ATTN(S)                                                        \
  extern "C" __global__ void k_attn_fwd_##S(                                   \
      const unsigned short*, unsigned short*, float*, int, int,                \
      const unsigned char*);                                                   \
  extern "C" __global__ void k_attn_fwd_drop_##S(                              \
      const unsigned short*, unsigned short*, float*, int, int,                \
      const unsigned char*, const unsigned long long*, float);                 \
  extern "C" __global__ void k_attn_bwd_q_##S(                                 \
      const unsigned short*, const unsigned short*, const unsigned short*,     \
      const float*, float*, unsigned short*, int, int, int,                    \
      const unsigned char*);                                                   \
  extern "C" __global__ void k_attn_bwd_q_drop_##S(                            \
      const unsigned short*, const unsigned short*, const unsigned short*,     \
      const float*, float*, unsigned short*, int, int, int,                    \
      const unsigned char*, const unsigned long long*, float);                 \
  extern "C" __global__ void k_attn_bwd_kv_##S(                                \
      const unsigned short*, const unsigned short*, const float*,              \
      const float*, unsigned short*, int, int, const unsigned char*);          \
  extern "C" __global__ void k_attn_bwd_kv_drop_##S(                           \
      const unsigned short*, const unsigned short*, const float*,              \
      const float*, unsigned short*, int, int, const unsigned char*,           \
      const unsigned long long*, float);
GA_DECL_ATTN(32) GA_DECL_ATTN(64) GA_DECL_ATTN(96) GA_DECL_ATTN(128)
extern "C" __global__ void k_attn_fwd4_128(
    const unsigned short*, unsigned short*, float*, int, int,
    const unsigned char*);
extern "C" __global__ void k_attn_fwd4_drop_128(
    const unsigned short*, unsigned short*, float*, int, int,
    const unsigned char*, const unsigned long long*, float);
extern "C" __global__ void k_attn_bwd8_128(
    const unsigned short*, const unsigned short*, const unsigned short*,
    const float*, unsigned short*, int, int, const unsigned char*);
extern "C" __global__ void k_attn_bwd8_drop_128(
    const unsigned short*, const unsigned short*, const unsigned short*,
    const float*, unsigned short*, int, int, const unsigned char*,
    const unsigned long long*, float);
extern "C" __global__ void k_attn_fwd_big4(
    const unsigned short*, unsigned short*, float*, int, int, int,
    const unsigned char*);
extern "C" __global__ void k_attn_fwd_big4_drop(
    const unsigned short*, unsigned short*, float*, int, int, int,
    const unsigned char*, const unsigned long long*, float);
extern "C" __global__ void k_attn_bwd_q_big4(
    const unsigned short*, const unsigned short*, const unsigned short*,
    const float*, float*, unsigned short*, int, int, int, int,
    const unsigned char*);
extern "C" __global__ void k_attn_bwd_q_big4_drop(
    const unsigned short*, const unsigned short*, const unsigned short*,
    const float*, float*, unsigned short*, int, int, int, int,
    const unsigned char*, const unsigned long long*, float);
extern "C" __global__ void k_attn_bwd_kv_big4(
    const unsigned short*, const unsigned short*, const float*,
    const float*, unsigned short*, int, int, int, const unsigned char*);
extern "C" __global__ void k_attn_bwd_kv_big4_drop(
    const unsigned short*, const unsigned short*, const float*,
    const float*, unsigned short*, int, int, int, const unsigned char*,
    const unsigned long long*, float);
// 4-wave S=128 path A/B (one workgroup per (b,h), panels staged once)
static inline bool attn_w4_on() {
  static const bool on = [] {
    const char* v = getenv("GA_ATTN_W4");
    return !v || atoi(v) != 0;
  }();
  return on;
}
extern "C" __global__ void k_attn_fwd_big(const unsigned short*, unsigned short*,
                                           float*, int, int, int,
                                           const unsigned char*);
extern "C" __global__ void k_attn_fwd_big_drop(const unsigned short*, unsigned short*,
                                               float*, int, int, int,
                                               const unsigned char*,
                                               const unsigned long long*, float);
extern "C" __global__ void k_attn_bwd_q_big(const unsigned short*, const unsigned short*,
                                            const unsigned short*, const float*,
                                            float*, unsigned short*, int, int, int, int,
                                            const unsigned char*);
extern "C" __global__ void k_attn_bwd_q_big_drop(
    const unsigned short*, const unsigned short*, const unsigned short*,
    const float*, float*, unsigned short*, int, int, int, int,
    const unsigned char*, const unsigned long long*, float);
extern "C" __global__ void k_attn_bwd_kv_big(const unsigned short*, const unsigned short*,
                                             const float*, const float*,
                                             unsigned short*, int, int, int,
                                             const unsigned char*);
extern "C" __global__ void k_attn_bwd_kv_big_drop(
    const unsigned short*, const unsigned short*, const float*, const float*,
    unsigned short*, int, int, int, const unsigned char*,
    const unsigned long long*, float);
extern "C" __global__ void k_attn_bwd_d(const unsigned short*, const unsigned short*,
                                        float*, int, int, int);
#define WG_MAX_G 24
struct WgArgs {
  unsigned long long x[WG_MAX_G];
  unsigned long long dy[WG_MAX_G];
  unsigned long long acc[WG_MAX_G];
  unsigned long long dbias[WG_MAX_G];
  int nk[WG_MAX_G * 2];
  int G;
};
extern "C" __global__ void k_wgrad_mfma(WgArgs, int);
extern "C" __global__ void k_wgrad_mfma256(WgArgs, int, int, float*, int*);
extern "C" __global__ void k_cls_head_fwd(const unsigned short*, const unsigned short*,
                                          const unsigned short*, const long long*,
                                          unsigned short*, float*, float*, int, int, int);
extern "C" __global__ void k_cls_head_bwd(const float*, const unsigned short*,
                                          const float*, const long long*,
                                          const unsigned short*, unsigned short*,
                                          float*, float*, int, int, int);

namespace {

inline void check_flat(const at::Tensor& t, const char* name, at::ScalarType dt) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
  TORCH_CHECK(t.scalar_type() == dt, name, " has wrong dtype");
  TORCH_CHECK(t.numel() % 64 == 0, name, " length must be 64-element aligned");
}

inline dim3 grid_for(long long n4) {
  long long b = (n4 + GA_THREADS - 1) / GA_THREADS;
  if (b > GA_MAX_BLOCKS) b = GA_MAX_BLOCKS;
  if (b < 1) b = 1;
  return dim3((unsigned)b);
}

void accumulate(at::Tensor accum, at::Tensor grads) {
  check_flat(accum, "accum", at::kFloat);
  TORCH_CHECK(grads.numel() == accum.numel(), "accum/grads size mismatch");
  const long long n4 = accum.numel() / 4;
  auto stream = c10::hip::getCurrentHIPStream().stream();
  if (grads.scalar_type() == at::kFloat) {
    hipLaunchKernelGGL(k_accum_f32, grid_for(n4), dim3(GA_THREADS), 0, stream,
                       (float4*)accum.data_ptr<float>(),
                       (float4*)grads.data_ptr<float>(), n4);
  } else if (grads.scalar_type() == at::kBFloat16) {
    hipLaunchKernelGGL(k_accum_bf16, grid_for(n4), dim3(GA_THREADS), 0, stream,
                       (float4*)accum.data_ptr<float>(),
                       (ushort4*)grads.data_ptr(), n4);
  } else {
    TORCH_CHECK(false, "grads must be fp32 or bf16");
  }
  TORCH_CHECK(hipGetLastError() == hipSuccess, "k_accum launch failed");
}

void sqnorm(at::Tensor accum, at::Tensor out) {
  check_flat(accum, "accum", at::kFloat);
  TORCH_CHECK(out.is_cuda() && out.scalar_type() == at::kFloat && out.numel() >= 1,
              "out must be a fp32 device scalar");
  const long long n4 = accum.numel() / 4;
  auto stream = c10::hip::getCurrentHIPStream().stream();
  hipMemsetAsync(out.data_ptr<float>(), 0, sizeof(float), stream);
  hipLaunchKernelGGL(k_sqnorm, grid_for(n4), dim3(GA_THREADS), 0, stream,
                     (const float4*)accum.data_ptr<float>(), n4,
                     out.data_ptr<float>());
  TORCH_CHECK(hipGetLastError() == hipSuccess, "k_sqnorm launch failed");
}

void fused_apply(at::Tensor accum, at::Tensor m, at::Tensor v, at::Tensor master,
                 at::Tensor model, bool has_model,
                 at::Tensor lr_dev, at::Tensor sqnorm_ws,
                 int64_t decay_boundary, double inv_k, double clip,
                 double weight_decay, double beta1, double beta2, double eps,
                 bool skip_norm = false) {
  // skip_norm: sqnorm_ws already holds the GLOBAL squared norm (sharded DP
  // apply: per-rank shard sqnorm + scalar all-reduce) -- do not recompute
  check_flat(accum, "accum", at::kFloat);
  check_flat(m, "m", at::kFloat);
  check_flat(v, "v", at::kFloat);
  check_flat(master, "master", at::kFloat);
  const long long n = accum.numel();
  TORCH_CHECK(m.numel() == n && v.numel() == n && master.numel() == n,
              "flat buffer size mismatch");
  TORCH_CHECK(decay_boundary % 64 == 0 && decay_boundary <= n, "bad decay boundary");
  const long long n4 = n / 4;
  auto stream = c10::hip::getCurrentHIPStream().stream();

  if (clip > 0.0 && !skip_norm) {
    hipMemsetAsync(sqnorm_ws.data_ptr<float>(), 0, sizeof(float), stream);
    hipLaunchKernelGGL(k_sqnorm, grid_for(n4), dim3(GA_THREADS), 0, stream,
                       (const float4*)accum.data_ptr<float>(), n4,
                       sqnorm_ws.data_ptr<float>());
  }
  if (has_model) {
    TORCH_CHECK(model.scalar_type() == at::kBFloat16 && model.numel() == n,
                "model buffer must be flat bf16 of same length");
    hipLaunchKernelGGL(apply_g2() ? k_apply_bf16_g2 : k_apply_bf16,
                       grid_for(n4), dim3(GA_THREADS), 0, stream,
                       (float4*)accum.data_ptr<float>(), (float4*)m.data_ptr<float>(),
                       (float4*)v.data_ptr<float>(), (float4*)master.data_ptr<float>(),
                       (ushort4*)model.data_ptr(),
                       lr_dev.data_ptr<float>(), sqnorm_ws.data_ptr<float>(),
                       n4, decay_boundary / 4,
                       (float)inv_k, (float)clip, (float)weight_decay,
                       (float)beta1, (float)beta2, (float)eps);
  } else {
    hipLaunchKernelGGL(apply_g2() ? k_apply_f32_g2 : k_apply_f32,
                       grid_for(n4), dim3(GA_THREADS), 0, stream,
                       (float4*)accum.data_ptr<float>(), (float4*)m.data_ptr<float>(),
                       (float4*)v.data_ptr<float>(), (float4*)master.data_ptr<float>(),
                       lr_dev.data_ptr<float>(), sqnorm_ws.data_ptr<float>(),
                       n4, decay_boundary / 4,
                       (float)inv_k, (float)clip, (float)weight_decay,
                       (float)beta1, (float)beta2, (float)eps);
  }
  TORCH_CHECK(hipGetLastError() == hipSuccess, "k_apply launch failed");
}

// ---------------- fused LN / GELU ----------------

inline const unsigned short* bfp(const at::Tensor& t) {
  return (const unsigned short*)t.data_ptr();
}
inline unsigned short* bfp_mut(at::Tensor& t) { return (unsigned short*)t.data_ptr(); }

inline void check_bf16_2d(const at::Tensor& t, const char* name, int H) {
  TORCH_CHECK(t.is_cuda() && t.is_contiguous() && t.scalar_type() == at::kBFloat16,
              name, " must be contiguous bf16 on GPU");
  TORCH_CHECK(t.numel() % H == 0, name, " numel not divisible by H");
}

std::vector<at::Tensor> addln_fwd(at::Tensor x, c10::optional<at::Tensor> res,
                                  c10::optional<at::Tensor> bias,
                                  at::Tensor gamma, at::Tensor beta, double eps) {
  const int H = (int)gamma.numel();
  TORCH_CHECK(H % 256 == 0 && H <= 1024, "LN hidden must be %256==0 and <=1024");
  check_bf16_2d(x, "x", H);
  const int R = (int)(x.numel() / H);
  auto y = at::empty_like(x);
  auto h = at::empty_like(x);
  auto mean = at::empty({R}, x.options().dtype(at::kFloat));
  auto rstd = at::empty({R}, x.options().dtype(at::kFloat));
  auto stream = c10::hip::getCurrentHIPStream().stream();
  // Split-row (2 waves/row) was measured SLOWER end-to-end on MI355X
  // (9752 vs 9965 samples/s): the per-row LDS exchanges + barriers cost
  // more than the extra occupancy buys. GA_LN_SPLIT2=1 re-enables it for
  // future re-measurement; kernels kept (fused_ln_gelu.hip SPLIT param).
  static const bool want_split = std::getenv("GA_LN_SPLIT2") != nullptr;
  const bool split2 = want_split && (H == 512 || H == 1024) && R >= 256;
  const int units = split2 ? R * 2 : R;
  // GA_LN_FWD_BLOCKS: like the backward's 256-cap (serial rows/wave
  // amortize the gamma/beta loads), A/B'd on the forward
  static const int fwd_cap = [] {
    const char* v = getenv("GA_LN_FWD_BLOCKS");
    return v ? atoi(v) : 2048;
  }();
  int blocks = std::max(1, std::min((units + 3) / 4, fwd_cap));
  void (*kern)(const unsigned short*, const unsigned short*, const unsigned short*,
               const unsigned short*, const unsigned short*, unsigned short*,
               unsigned short*, float*, float*, int, int, float) =
      split2 ? (H == 512 ? k_addln_fwd_s2_4 : k_addln_fwd_s2_8)
      : H == 256 ? k_addln_fwd_4 : H == 512 ? k_addln_fwd_8
      : H == 768 ? k_addln_fwd_12 : k_addln_fwd_16;
  hipLaunchKernelGGL(kern, dim3(blocks), dim3(256), 0, stream,
                     bfp(x), res ? bfp(*res) : nullptr, bias ? bfp(*bias) : nullptr,
                     bfp(gamma), bfp(beta), bfp_mut(y), bfp_mut(h),
                     mean.data_ptr<float>(), rstd.data_ptr<float>(), R, H, (float)eps);
  TORCH_CHECK(hipGetLastError() == hipSuccess, "k_addln_fwd launch failed");
  return {y, h, mean, rstd};
}

std::vector<at::Tensor> addln_bwd(at::Tensor dy, at::Tensor h, at::Tensor gamma,
                                  at::Tensor mean, at::Tensor rstd) {
  const int H = (int)gamma.numel();
  check_bf16_2d(dy, "dy", H);
  const int R = (int)(dy.numel() / H);
  auto dh = at::empty_like(dy);
  static const bool want_split = std::getenv("GA_LN_SPLIT2") != nullptr;
  const bool split2 = want_split && (H == 512 || H == 1024) && R >= 256;
  const int units = split2 ? R * 2 : R;
  // block cap: GA_LN_BWD_BLOCKS overrides for A/B (256 = one 4-wave
  // workgroup per CU with 4 serial rows/wave at fused R=4096; 1024 = one
  // row per wave, 4x the partial slabs)
  static const int bwd_cap = [] {
    const char* v = getenv("GA_LN_BWD_BLOCKS");
    return v ? atoi(v) : 256;
  }();
  int blocks = std::max(1, std::min((units + 3) / 4, bwd_cap));
  auto partials = at::empty({blocks, 3, H}, dy.options().dtype(at::kFloat));
  auto stream = c10::hip::getCurrentHIPStream().stream();
  size_t lds = (size_t)4 * 3 * (split2 ? H / 2 : H) * sizeof(float);
  void (*kern)(const unsigned short*, const unsigned short*, const unsigned short*,
               const float*, const float*, unsigned short*, float*, int, int) =
      split2 ? (H == 512 ? k_addln_bwd_s2_4 : k_addln_bwd_s2_8)
      : H == 256 ? k_addln_bwd_4 : H == 512 ? k_addln_bwd_8
      : H == 768 ? k_addln_bwd_12 : k_addln_bwd_16;
  hipLaunchKernelGGL(kern, dim3(blocks), dim3(256), lds, stream,
                     bfp(dy), bfp(h), bfp(gamma), mean.data_ptr<float>(),
                     rstd.data_ptr<float>(), bfp_mut(dh),
                     partials.data_ptr<float>(), R, H);
  TORCH_CHECK(hipGetLastError() == hipSuccess, "k_addln_bwd launch failed");
  return {dh, partials};
}

at::Tensor biasgelu_fwd(at::Tensor x, at::Tensor bias) {
  const int H = (int)bias.numel();
  TORCH_CHECK(H % 256 == 0 && H <= 4096, "gelu hidden must be %256==0 and <=4096");
  check_bf16_2d(x, "x", H);
  auto y = at::empty_like(x);
  const long long total = x.numel();
  const int R = (int)(total / H);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  // 2D: (column chunks of 2048 elems) x (rows); no per-thread modulo
  int cblocks = (H / 8 + 255) / 256;
  hipLaunchKernelGGL(k_biasgelu_fwd, dim3(cblocks, R), dim3(256), 0, stream,
                     bfp(x), bfp(bias), bfp_mut(y), total, H);
  TORCH_CHECK(hipGetLastError() == hipSuccess, "k_biasgelu_fwd launch failed");
  return y;
}

std::vector<at::Tensor> biasgelu_bwd(at::Tensor dy, at::Tensor x, at::Tensor bias) {
  const int H = (int)bias.numel();
  check_bf16_2d(dy, "dy", H);
  const int R = (int)(dy.numel() / H);
  auto dx = at::empty_like(dy);
  int blocks = std::max(1, std::min((R + 3) / 4, 256));
  auto partials = at::empty({blocks, H}, dy.options().dtype(at::kFloat));
  auto stream = c10::hip::getCurrentHIPStream().stream();
  size_t lds = (size_t)4 * H * sizeof(float);
  void (*kern)(const unsigned short*, const unsigned short*, const unsigned short*,
               unsigned short*, float*, int, int) =
      H == 512 ? k_biasgelu_bwd_8 : H == 768 ? k_biasgelu_bwd_12
      : H == 1024 ? k_biasgelu_bwd_16 : H == 1536 ? k_biasgelu_bwd_24
      : H == 2048 ? k_biasgelu_bwd_32 : H == 3072 ? k_biasgelu_bwd_48
                                                  : k_biasgelu_bwd_64;
  TORCH_CHECK(H == 512 || H == 768 || H == 1024 || H == 1536 || H == 2048 ||
                  H == 3072 || H == 4096,
              "gelu hidden must be one of 512/768/1024/1536/2048/3072/4096");
  hipLaunchKernelGGL(kern, dim3(blocks), dim3(256), lds, stream,
                     bfp(dy), bfp(x), bfp(bias), bfp_mut(dx),
                     partials.data_ptr<float>(), R, H);
  TORCH_CHECK(hipGetLastError() == hipSuccess, "k_biasgelu_bwd launch failed");
  return {dx, partials};
}

void colreduce_acc(at::Tensor partials, at::Tensor dest0,
                   c10::optional<at::Tensor> dest1, c10::optional<at::Tensor> dest2) {
  TORCH_CHECK(partials.dim() >= 2 && partials.scalar_type() == at::kFloat &&
              partials.is_contiguous(), "partials must be contiguous fp32 [NB,...]");
  const int NB = (int)partials.size(0);
  const int C = (int)(partials.numel() / NB);
  const int n0 = (int)dest0.numel();
  const int n1 = dest1 ? (int)dest1->numel() : 0;
  const int n2 = dest2 ? (int)dest2->numel() : 0;
  TORCH_CHECK(n0 + n1 + n2 == C, "dest sizes must sum to partial columns");
  auto stream = c10::hip::getCurrentHIPStream().stream();
  const int nch = (NB + 7) / 8;
  int blocks = (int)std::min<long long>(((long long)C * nch + 255) / 256, 2048);
  hipLaunchKernelGGL(k_colreduce_acc, dim3(blocks), dim3(256), 0, stream,
                     partials.data_ptr<float>(), NB, C,
                     dest0.data_ptr<float>(), n0,
                     dest1 ? dest1->data_ptr<float>() : nullptr, n1,
                     dest2 ? dest2->data_ptr<float>() : nullptr);
  TORCH_CHECK(hipGetLastError() == hipSuccess, "k_colreduce_acc launch failed");
}

// qkv [B,S,3,H] bf16 -> (out [B,S,H], lse [B,nh,S] fp32). S<=128, S%32==0,
// head_dim 64 (all reference BERT configs).
// NOTE on a measured negative result: forking bwd_kv onto a side stream
// (bwd_q and bwd_kv are independent once the D table exists from
// k_attn_bwd_d) regressed the bench 10390 -> 8286 samples/s -- inside the
// captured graph the extra event edges + the standalone D kernel cost far
// more than the ~4 us/layer of overlap they buy. The sequential launch
// below (bwd_q computes and publishes D itself) is the fast path.
static const unsigned char* mask_ptr_of(const c10::optional<at::Tensor>& mask,
                                        int B, int S) {
  if (!mask.has_value()) return nullptr;
  const auto& m = mask.value();
  TORCH_CHECK(m.is_cuda() && m.is_contiguous() && m.scalar_type() == at::kByte,
              "attention mask must be contiguous u8 on device");
  TORCH_CHECK(m.dim() == 2 && m.size(0) == B && m.size(1) == S,
              "attention mask must be [B,S]");
  return (const unsigned char*)m.data_ptr();
}

static const unsigned long long* seed_ptr_of(const c10::optional<at::Tensor>& seed) {
  if (!seed.has_value()) return nullptr;
  const auto& s = seed.value();
  TORCH_CHECK(s.is_cuda() && s.scalar_type() == at::kLong && s.numel() == 1,
              "dropout seed must be a device int64 scalar");
  return (const unsigned long long*)s.data_ptr();
}

std::vector<at::Tensor> attn_fwd(at::Tensor qkv, int64_t nh,
                                 c10::optional<at::Tensor> mask,
                                 c10::optional<at::Tensor> seed, double p_drop) {
  TORCH_CHECK(qkv.is_cuda() && qkv.is_contiguous() &&
              qkv.scalar_type() == at::kBFloat16, "qkv must be contiguous bf16");
  TORCH_CHECK(qkv.dim() == 4 && qkv.size(2) == 3, "qkv must be [B,S,3,H]");
  const int B = (int)qkv.size(0), S = (int)qkv.size(1), H = (int)qkv.size(3);
  TORCH_CHECK(H == nh * 64, "head_dim must be 64");
  TORCH_CHECK((S <= 128 && S % 32 == 0) || S % 64 == 0,
              "attn kernel needs S%32==0 and (S<=128 or S%64==0)");
  const bool drop = p_drop > 0.0;
  TORCH_CHECK(!drop || seed.has_value(), "dropout needs a device seed scalar");
  const unsigned char* mp = mask_ptr_of(mask, B, S);
  const unsigned long long* sp = seed_ptr_of(seed);
  const float pd = (float)p_drop;
  auto out = at::empty({B, S, H}, qkv.options());
  auto lse = at::empty({B, nh, S}, qkv.options().dtype(at::kFloat));
  auto stream = c10::hip::getCurrentHIPStream().stream();
  const int NT = S / 32;
  if (S > 128) {
    // chunked online-softmax variants: [64][64] K + V^T LDS panels (+mask).
    // Default 4-wave grouping (chunks staged once per four q-tile waves);
    // GA_ATTN_W4=0 reverts to the 1-wave-per-tile kernels.
    if (attn_w4_on()) {
      const int ngrp = (NT + 3) / 4;
      if (drop)
        hipLaunchKernelGGL(k_attn_fwd_big4_drop, dim3(B * (int)nh * ngrp),
                           dim3(256), 16384 + 256, stream,
                           (const unsigned short*)qkv.data_ptr(),
                           (unsigned short*)out.data_ptr(), lse.data_ptr<float>(),
                           B, S, (int)nh, mp, sp, pd);
      else
        hipLaunchKernelGGL(k_attn_fwd_big4, dim3(B * (int)nh * ngrp),
                           dim3(256), 16384 + 256, stream,
                           (const unsigned short*)qkv.data_ptr(),
                           (unsigned short*)out.data_ptr(), lse.data_ptr<float>(),
                           B, S, (int)nh, mp);
    } else if (drop)
      hipLaunchKernelGGL(k_attn_fwd_big_drop, dim3(B * (int)nh * NT), dim3(64),
                         16384 + 256, stream,
                         (const unsigned short*)qkv.data_ptr(),
                         (unsigned short*)out.data_ptr(), lse.data_ptr<float>(),
                         B, S, (int)nh, mp, sp, pd);
    else
      hipLaunchKernelGGL(k_attn_fwd_big, dim3(B * (int)nh * NT), dim3(64),
                         16384 + 256, stream,
                         (const unsigned short*)qkv.data_ptr(),
                         (unsigned short*)out.data_ptr(), lse.data_ptr<float>(),
                         B, S, (int)nh, mp);
    TORCH_CHECK(hipGetLastError() == hipSuccess, "k_attn_fwd_big launch failed");
    return {out, lse};
  }
  if (S == 128 && attn_w4_on()) {
    const size_t lds4 = 16384 + 16384 + 512;
    if (drop)
      hipLaunchKernelGGL(k_attn_fwd4_drop_128, dim3(B * (int)nh), dim3(256),
                         lds4, stream, (const unsigned short*)qkv.data_ptr(),
                         (unsigned short*)out.data_ptr(), lse.data_ptr<float>(),
                         B, (int)nh, mp, sp, pd);
    else
      hipLaunchKernelGGL(k_attn_fwd4_128, dim3(B * (int)nh), dim3(256),
                         lds4, stream, (const unsigned short*)qkv.data_ptr(),
                         (unsigned short*)out.data_ptr(), lse.data_ptr<float>(),
                         B, (int)nh, mp);
    TORCH_CHECK(hipGetLastError() == hipSuccess, "k_attn_fwd4 launch failed");
    return {out, lse};
  }
  const size_t lds = 16384 + 64 * 256 + 512;  // K + V^T panels + mask table
  if (drop) {
    void (*fk)(const unsigned short*, unsigned short*, float*, int, int,
               const unsigned char*, const unsigned long long*, float) =
        S == 32 ? k_attn_fwd_drop_32 : S == 64 ? k_attn_fwd_drop_64
        : S == 96 ? k_attn_fwd_drop_96 : k_attn_fwd_drop_128;
    hipLaunchKernelGGL(fk, dim3(B * (int)nh * NT), dim3(64), lds, stream,
                       (const unsigned short*)qkv.data_ptr(),
                       (unsigned short*)out.data_ptr(), lse.data_ptr<float>(),
                       B, (int)nh, mp, sp, pd);
  } else {
    void (*fk)(const unsigned short*, unsigned short*, float*, int, int,
               const unsigned char*) =
        S == 32 ? k_attn_fwd_32 : S == 64 ? k_attn_fwd_64
        : S == 96 ? k_attn_fwd_96 : k_attn_fwd_128;
    hipLaunchKernelGGL(fk, dim3(B * (int)nh * NT), dim3(64), lds, stream,
                       (const unsigned short*)qkv.data_ptr(),
                       (unsigned short*)out.data_ptr(), lse.data_ptr<float>(),
                       B, (int)nh, mp);
  }
  TORCH_CHECK(hipGetLastError() == hipSuccess, "k_attn_fwd launch failed");
  return {out, lse};
}

at::Tensor attn_bwd(at::Tensor qkv, at::Tensor out, at::Tensor dout,
                    at::Tensor lse, int64_t nh,
                    c10::optional<at::Tensor> mask,
                    c10::optional<at::Tensor> seed, double p_drop) {
  const int B = (int)qkv.size(0), S = (int)qkv.size(1), H = (int)qkv.size(3);
  TORCH_CHECK(dout.is_contiguous() && dout.scalar_type() == at::kBFloat16);
  const bool drop = p_drop > 0.0;
  TORCH_CHECK(!drop || seed.has_value(), "dropout needs a device seed scalar");
  const unsigned char* mp = mask_ptr_of(mask, B, S);
  const unsigned long long* sp = seed_ptr_of(seed);
  const float pd = (float)p_drop;
  auto dqkv = at::empty_like(qkv);
  auto Dtab = at::empty({B, (long)nh, S}, lse.options());
  auto stream = c10::hip::getCurrentHIPStream().stream();
  const int NT = S / 32;

  if (S > 128 && attn_w4_on()) {
    const int ngrp = (NT + 3) / 4;
    if (drop)
      hipLaunchKernelGGL(k_attn_bwd_q_big4_drop, dim3(B * (int)nh * ngrp),
                         dim3(256), 24576 + 256, stream,
                         (const unsigned short*)qkv.data_ptr(),
                         (const unsigned short*)out.data_ptr(),
                         (const unsigned short*)dout.data_ptr(),
                         lse.data_ptr<float>(), Dtab.data_ptr<float>(),
                         (unsigned short*)dqkv.data_ptr(), B, S, (int)nh, 1,
                         mp, sp, pd);
    else
      hipLaunchKernelGGL(k_attn_bwd_q_big4, dim3(B * (int)nh * ngrp),
                         dim3(256), 24576 + 256, stream,
                         (const unsigned short*)qkv.data_ptr(),
                         (const unsigned short*)out.data_ptr(),
                         (const unsigned short*)dout.data_ptr(),
                         lse.data_ptr<float>(), Dtab.data_ptr<float>(),
                         (unsigned short*)dqkv.data_ptr(), B, S, (int)nh, 1, mp);
    if (drop)
      hipLaunchKernelGGL(k_attn_bwd_kv_big4_drop, dim3(B * (int)nh * ngrp),
                         dim3(256), 16384 + 512, stream,
                         (const unsigned short*)qkv.data_ptr(),
                         (const unsigned short*)dout.data_ptr(),
                         lse.data_ptr<float>(), Dtab.data_ptr<float>(),
                         (unsigned short*)dqkv.data_ptr(), B, S, (int)nh,
                         mp, sp, pd);
    else
      hipLaunchKernelGGL(k_attn_bwd_kv_big4, dim3(B * (int)nh * ngrp),
                         dim3(256), 16384 + 512, stream,
                         (const unsigned short*)qkv.data_ptr(),
                         (const unsigned short*)dout.data_ptr(),
                         lse.data_ptr<float>(), Dtab.data_ptr<float>(),
                         (unsigned short*)dqkv.data_ptr(), B, S, (int)nh, mp);
  } else if (S > 128) {
    if (drop)
      hipLaunchKernelGGL(k_attn_bwd_q_big_drop, dim3(B * (int)nh * NT), dim3(64),
                         24576 + 256, stream,
                         (const unsigned short*)qkv.data_ptr(),
                         (const unsigned short*)out.data_ptr(),
                         (const unsigned short*)dout.data_ptr(),
                         lse.data_ptr<float>(), Dtab.data_ptr<float>(),
                         (unsigned short*)dqkv.data_ptr(), B, S, (int)nh, 1,
                         mp, sp, pd);
    else
      hipLaunchKernelGGL(k_attn_bwd_q_big, dim3(B * (int)nh * NT), dim3(64),
                         24576 + 256, stream,
                         (const unsigned short*)qkv.data_ptr(),
                         (const unsigned short*)out.data_ptr(),
                         (const unsigned short*)dout.data_ptr(),
                         lse.data_ptr<float>(), Dtab.data_ptr<float>(),
                         (unsigned short*)dqkv.data_ptr(), B, S, (int)nh, 1, mp);
    if (drop)
      hipLaunchKernelGGL(k_attn_bwd_kv_big_drop, dim3(B * (int)nh * NT), dim3(64),
                         16384 + 512, stream,
                         (const unsigned short*)qkv.data_ptr(),
                         (const unsigned short*)dout.data_ptr(),
                         lse.data_ptr<float>(), Dtab.data_ptr<float>(),
                         (unsigned short*)dqkv.data_ptr(), B, S, (int)nh,
                         mp, sp, pd);
    else
      hipLaunchKernelGGL(k_attn_bwd_kv_big, dim3(B * (int)nh * NT), dim3(64),
                         16384 + 512, stream,
                         (const unsigned short*)qkv.data_ptr(),
                         (const unsigned short*)dout.data_ptr(),
                         lse.data_ptr<float>(), Dtab.data_ptr<float>(),
                         (unsigned short*)dqkv.data_ptr(), B, S, (int)nh, mp);
  } else if (S == 128 && attn_w4_on()) {
    // merged 8-wave backward: one launch per layer, D table through LDS
    const size_t lds8 = 16384 * 5 + 3 * 512;
    if (drop)
      hipLaunchKernelGGL(k_attn_bwd8_drop_128, dim3(B * (int)nh), dim3(512),
                         lds8, stream,
                         (const unsigned short*)qkv.data_ptr(),
                         (const unsigned short*)out.data_ptr(),
                         (const unsigned short*)dout.data_ptr(),
                         lse.data_ptr<float>(),
                         (unsigned short*)dqkv.data_ptr(), B, (int)nh,
                         mp, sp, pd);
    else
      hipLaunchKernelGGL(k_attn_bwd8_128, dim3(B * (int)nh), dim3(512),
                         lds8, stream,
                         (const unsigned short*)qkv.data_ptr(),
                         (const unsigned short*)out.data_ptr(),
                         (const unsigned short*)dout.data_ptr(),
                         lse.data_ptr<float>(),
                         (unsigned short*)dqkv.data_ptr(), B, (int)nh, mp);
  } else {
    const size_t lds_q = 16384 * 3 + 512;  // K + V + K^T + mask table
    if (drop) {
      void (*qk)(const unsigned short*, const unsigned short*, const unsigned short*,
                 const float*, float*, unsigned short*, int, int, int,
                 const unsigned char*, const unsigned long long*, float) =
          S == 32 ? k_attn_bwd_q_drop_32 : S == 64 ? k_attn_bwd_q_drop_64
          : S == 96 ? k_attn_bwd_q_drop_96 : k_attn_bwd_q_drop_128;
      hipLaunchKernelGGL(qk, dim3(B * (int)nh * NT), dim3(64), lds_q, stream,
                         (const unsigned short*)qkv.data_ptr(),
                         (const unsigned short*)out.data_ptr(),
                         (const unsigned short*)dout.data_ptr(),
                         lse.data_ptr<float>(), Dtab.data_ptr<float>(),
                         (unsigned short*)dqkv.data_ptr(), B, (int)nh, 1,
                         mp, sp, pd);
    } else {
      void (*qk)(const unsigned short*, const unsigned short*, const unsigned short*,
                 const float*, float*, unsigned short*, int, int, int,
                 const unsigned char*) =
          S == 32 ? k_attn_bwd_q_32 : S == 64 ? k_attn_bwd_q_64
          : S == 96 ? k_attn_bwd_q_96 : k_attn_bwd_q_128;
      hipLaunchKernelGGL(qk, dim3(B * (int)nh * NT), dim3(64), lds_q, stream,
                         (const unsigned short*)qkv.data_ptr(),
                         (const unsigned short*)out.data_ptr(),
                         (const unsigned short*)dout.data_ptr(),
                         lse.data_ptr<float>(), Dtab.data_ptr<float>(),
                         (unsigned short*)dqkv.data_ptr(), B, (int)nh, 1, mp);
    }
    const size_t lds_kv = (size_t)64 * 256 * 2 + 1024;  // dO^T + Q^T + tables
    if (drop) {
      void (*kvk)(const unsigned short*, const unsigned short*, const float*,
                  const float*, unsigned short*, int, int, const unsigned char*,
                  const unsigned long long*, float) =
          S == 32 ? k_attn_bwd_kv_drop_32 : S == 64 ? k_attn_bwd_kv_drop_64
          : S == 96 ? k_attn_bwd_kv_drop_96 : k_attn_bwd_kv_drop_128;
      hipLaunchKernelGGL(kvk, dim3(B * (int)nh * NT), dim3(64), lds_kv, stream,
                         (const unsigned short*)qkv.data_ptr(),
                         (const unsigned short*)dout.data_ptr(),
                         lse.data_ptr<float>(), Dtab.data_ptr<float>(),
                         (unsigned short*)dqkv.data_ptr(), B, (int)nh,
                         mp, sp, pd);
    } else {
      void (*kvk)(const unsigned short*, const unsigned short*, const float*,
                  const float*, unsigned short*, int, int, const unsigned char*) =
          S == 32 ? k_attn_bwd_kv_32 : S == 64 ? k_attn_bwd_kv_64
          : S == 96 ? k_attn_bwd_kv_96 : k_attn_bwd_kv_128;
      hipLaunchKernelGGL(kvk, dim3(B * (int)nh * NT), dim3(64), lds_kv, stream,
                         (const unsigned short*)qkv.data_ptr(),
                         (const unsigned short*)dout.data_ptr(),
                         lse.data_ptr<float>(), Dtab.data_ptr<float>(),
                         (unsigned short*)dqkv.data_ptr(), B, (int)nh, mp);
    }
  }
  TORCH_CHECK(hipGetLastError() == hipSuccess, "k_attn_bwd launch failed");
  return dqkv;
}

// One launch over every pending wgrad; metadata passed by value (see
// wgrad_mfma.hip) so it is hipGraph-capture-safe.
void wgrad_mfma(std::vector<at::Tensor> xs, std::vector<at::Tensor> dys,
                std::vector<at::Tensor> accs, std::vector<at::Tensor> dbias,
                int64_t R) {
  const int G = (int)xs.size();
  TORCH_CHECK(G > 0 && G <= WG_MAX_G, "wgrad_mfma: 1..24 problems");
  TORCH_CHECK(dbias.empty() || dbias.size() == xs.size(), "dbias size");
  WgArgs args{};
  args.G = G;
  long long ntiles = 0;
  for (int g = 0; g < G; ++g) {
    const int K = (int)xs[g].size(-1), N = (int)dys[g].size(-1);
    TORCH_CHECK(N % 128 == 0 && K % 128 == 0, "wgrad_mfma: N,K % 128");
    TORCH_CHECK(xs[g].numel() / K == R && dys[g].numel() / N == R, "R mismatch");
    TORCH_CHECK(accs[g].numel() == (long long)N * K, "acc slice size");
    args.x[g] = (unsigned long long)xs[g].data_ptr();
    args.dy[g] = (unsigned long long)dys[g].data_ptr();
    args.acc[g] = (unsigned long long)accs[g].data_ptr();
    if (!dbias.empty() && dbias[g].numel() > 0) {
      TORCH_CHECK(dbias[g].numel() == N && dbias[g].scalar_type() == at::kFloat,
                  "dbias slice must be fp32 [N]");
      args.dbias[g] = (unsigned long long)dbias[g].data_ptr();
    }
    args.nk[g * 2 + 0] = N;
    args.nk[g * 2 + 1] = K;
    ntiles += (long long)(N / 128) * (K / 128);
  }
  bool all256 = true;
  for (int g = 0; g < G; ++g)
    all256 &= (args.nk[g * 2] % 256 == 0) && (args.nk[g * 2 + 1] % 256 == 0);
  // A/B knobs (re-measurement): GA_WGRAD_TILE=128 forces the 4-wave
  // 128-tile kernel even when 256 tiles fit; GA_WGRAD_SPLITS=N forces
  // R-splitting with the split-K-fixup combine (scratch partials +
  // last-arrival reduce). Both combines MEASURED SLOWER at the fused
  // bench shapes: scalar atomics 19.6k -> 8.4k (~51M serialized fp32
  // atomics/window), coalesced fixup 19.6k -> 13.4k at auto-splits=4 --
  // the kernel is operand-bandwidth-bound (panel re-reads ride L2), and
  // splitting multiplies the in-flight working set past what L2 holds.
  // Splits therefore default OFF; the machinery stays for re-measurement
  // on future parts.
  static const int env_tile = [] {
    const char* v = getenv("GA_WGRAD_TILE");
    return v ? atoi(v) : 256;
  }();
  static const int env_splits = [] {
    const char* v = getenv("GA_WGRAD_SPLITS");
    return v ? atoi(v) : -1;  // -1 = auto
  }();
  all256 &= env_tile == 256;
  auto stream = c10::hip::getCurrentHIPStream().stream();
  if (all256) {
    long long nt256 = 0;
    for (int g = 0; g < G; ++g)
      nt256 += (long long)(args.nk[g * 2] / 256) * (args.nk[g * 2 + 1] / 256);
    const long long chunks = R / 64;
    int splits = env_splits < 1 ? 1 : env_splits;
    while (splits > 1 && (long long)splits * 4 > chunks) splits /= 2;
    float* scratch = nullptr;
    int* counters = nullptr;
    if (splits > 1) {
      // cached device workspace (leaky holders: stable pointers for graph
      // replays, no static-destructor-vs-context teardown hazard).
      // Allocated during eager warmup; replays reuse the pointers.
      static auto* scratch_t = new at::Tensor();
      static auto* counters_t = new at::Tensor();
      const long long need = nt256 * (long long)splits * 65536;
      if (!scratch_t->defined() || scratch_t->numel() < need)
        *scratch_t = at::empty({need}, accs[0].options().dtype(at::kFloat));
      if (!counters_t->defined() || counters_t->numel() < nt256)
        *counters_t = at::zeros({nt256}, accs[0].options().dtype(at::kInt));
      scratch = scratch_t->data_ptr<float>();
      counters = counters_t->data_ptr<int>();
    }
    hipLaunchKernelGGL(k_wgrad_mfma256, dim3((unsigned)(nt256 * splits)),
                       dim3(512), 65536, stream, args, (int)R, splits,
                       scratch, counters);
  } else {
    hipLaunchKernelGGL(k_wgrad_mfma, dim3((unsigned)ntiles), dim3(256), 65536,
                       stream, args, (int)R);
  }
  TORCH_CHECK(hipGetLastError() == hipSuccess, "k_wgrad_mfma launch failed");
}

std::vector<at::Tensor> cls_head_fwd(at::Tensor pre, at::Tensor Wc, at::Tensor bc,
                                     at::Tensor labels) {
  const int B = (int)pre.size(0), H = (int)pre.size(1), C = (int)Wc.size(0);
  TORCH_CHECK(B <= 4096 && C <= 8 && H % 512 == 0 && H <= 1024,
              "cls head: B<=4096, C<=8, H in {512, 1024}");
  TORCH_CHECK(pre.is_contiguous() && pre.scalar_type() == at::kBFloat16);
  auto t = at::empty_like(pre);
  auto probs = at::empty({B, C}, pre.options().dtype(at::kFloat));
  auto loss = at::zeros({}, pre.options().dtype(at::kFloat));  // atomic target
  auto stream = c10::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(k_cls_head_fwd, dim3((B + 3) / 4), dim3(256), 0, stream,
                     (const unsigned short*)pre.data_ptr(),
                     (const unsigned short*)Wc.data_ptr(),
                     (const unsigned short*)bc.data_ptr(),
                     (const long long*)labels.data_ptr<int64_t>(),
                     (unsigned short*)t.data_ptr(), probs.data_ptr<float>(),
                     loss.data_ptr<float>(), B, H, C);
  TORCH_CHECK(hipGetLastError() == hipSuccess, "k_cls_head_fwd launch failed");
  return {loss, t, probs};
}

at::Tensor cls_head_bwd(at::Tensor dloss, at::Tensor t, at::Tensor probs,
                        at::Tensor labels, at::Tensor Wc,
                        c10::optional<at::Tensor> acc_w,
                        c10::optional<at::Tensor> acc_b) {
  const int B = (int)t.size(0), H = (int)t.size(1), C = (int)Wc.size(0);
  auto dpre = at::empty_like(t);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(k_cls_head_bwd,
                     dim3((H + 255) / 256, (B + 7) / 8), dim3(256), 0, stream,
                     dloss.data_ptr<float>(), (const unsigned short*)t.data_ptr(),
                     probs.data_ptr<float>(),
                     (const long long*)labels.data_ptr<int64_t>(),
                     (const unsigned short*)Wc.data_ptr(),
                     (unsigned short*)dpre.data_ptr(),
                     acc_w ? acc_w->data_ptr<float>() : nullptr,
                     acc_b ? acc_b->data_ptr<float>() : nullptr, B, H, C);
  TORCH_CHECK(hipGetLastError() == hipSuccess, "k_cls_head_bwd launch failed");
  return dpre;
}

// problems: list of (partials [NB,...,C], dest0, dest1?, dest2?)
// ---- small-GEMM MFMA Linear path (linear_small.hip) ----
at::Tensor emb3_fwd(at::Tensor ids, c10::optional<at::Tensor> tids,
                    at::Tensor word, at::Tensor pos,
                    c10::optional<at::Tensor> tok) {
  TORCH_CHECK(ids.is_cuda() && ids.scalar_type() == at::kLong && ids.dim() == 2);
  TORCH_CHECK(word.is_contiguous() && word.scalar_type() == at::kBFloat16);
  const int B = (int)ids.size(0), S = (int)ids.size(1);
  const int H = (int)word.size(1);
  TORCH_CHECK(H % 256 == 0, "emb3_fwd hidden must be %256==0");
  auto out = at::empty({B, S, H}, word.options());
  auto stream = c10::hip::getCurrentHIPStream().stream();
  int cblocks = (H / 8 + 255) / 256;
  auto idsc = ids.contiguous();
  const long long* tp = nullptr;
  at::Tensor tidsc;
  if (tids) {
    tidsc = tids->contiguous();
    tp = (const long long*)tidsc.data_ptr<int64_t>();
  }
  hipLaunchKernelGGL(k_emb3_fwd, dim3(cblocks, B * S), dim3(256), 0, stream,
                     (const long long*)idsc.data_ptr<int64_t>(), tp,
                     (const unsigned short*)word.data_ptr(),
                     (const unsigned short*)pos.data_ptr(),
                     tok ? (const unsigned short*)tok->data_ptr() : nullptr,
                     (unsigned short*)out.data_ptr(), S, H);
  TORCH_CHECK(hipGetLastError() == hipSuccess, "k_emb3_fwd launch failed");
  return out;
}

at::Tensor lin_fwd_small(at::Tensor x, at::Tensor w, c10::optional<at::Tensor> bias) {
  const int N = (int)w.size(0), K = (int)w.size(1);
  const int R = (int)(x.numel() / K);
  TORCH_CHECK(x.is_contiguous() && w.is_contiguous() && x.scalar_type() == at::kBFloat16,
              "lin_fwd_small: contiguous bf16 required");
  TORCH_CHECK(R % 64 == 0 && N % 64 == 0 && K % 64 == 0, "lin_fwd_small shape");
  auto sizes = x.sizes().vec();
  sizes.back() = N;
  auto y = at::empty(sizes, x.options());
  auto stream = c10::hip::getCurrentHIPStream().stream();
  const int tiles = (R / 64) * (N / 64);
  const int S = (K >= 1024) ? 4 : 1;
  const unsigned short* bp = bias ? (const unsigned short*)bias->data_ptr() : nullptr;
  if (S == 1) {
    hipLaunchKernelGGL(k_lin_fwd_bf16, dim3(tiles), dim3(64), 0, stream,
                       (const unsigned short*)x.data_ptr(),
                       (const unsigned short*)w.data_ptr(), bp,
                       (unsigned short*)y.data_ptr(), R, N, K);
  } else {
    auto part = at::empty({S, (long)R, (long)N}, x.options().dtype(at::kFloat));
    hipLaunchKernelGGL(k_lin_fwd_f32, dim3(tiles, S), dim3(64), 0, stream,
                       (const unsigned short*)x.data_ptr(),
                       (const unsigned short*)w.data_ptr(),
                       part.data_ptr<float>(), R, N, K, S);
    const long long total = (long long)R * N;
    int blocks = (int)std::min<long long>((total / 4 + 255) / 256, 2048);
    hipLaunchKernelGGL(k_splitk_combine, dim3(blocks), dim3(256), 0, stream,
                       part.data_ptr<float>(), S, total, N, bp,
                       (unsigned short*)y.data_ptr());
  }
  TORCH_CHECK(hipGetLastError() == hipSuccess, "lin_fwd_small launch failed");
  return y;
}

at::Tensor lin_dgrad_small(at::Tensor dy, at::Tensor w) {
  const int N = (int)w.size(0), K = (int)w.size(1);
  const int R = (int)(dy.numel() / N);
  TORCH_CHECK(dy.is_contiguous() && w.is_contiguous() && dy.scalar_type() == at::kBFloat16,
              "lin_dgrad_small: contiguous bf16 required");
  TORCH_CHECK(R % 64 == 0 && N % 64 == 0 && K % 64 == 0, "lin_dgrad_small shape");
  auto sizes = dy.sizes().vec();
  sizes.back() = K;
  auto dx = at::empty(sizes, dy.options());
  auto stream = c10::hip::getCurrentHIPStream().stream();
  const int tiles = (R / 64) * (K / 64);
  const int S = (N >= 1024) ? 4 : 1;
  if (S == 1) {
    hipLaunchKernelGGL(k_lin_dgrad_bf16, dim3(tiles), dim3(64), 0, stream,
                       (const unsigned short*)dy.data_ptr(),
                       (const unsigned short*)w.data_ptr(),
                       (unsigned short*)dx.data_ptr(), R, N, K);
  } else {
    auto part = at::empty({S, (long)R, (long)K}, dy.options().dtype(at::kFloat));
    hipLaunchKernelGGL(k_lin_dgrad_f32, dim3(tiles, S), dim3(64), 0, stream,
                       (const unsigned short*)dy.data_ptr(),
                       (const unsigned short*)w.data_ptr(),
                       part.data_ptr<float>(), R, N, K, S);
    const long long total = (long long)R * K;
    int blocks = (int)std::min<long long>((total / 4 + 255) / 256, 2048);
    hipLaunchKernelGGL(k_splitk_combine, dim3(blocks), dim3(256), 0, stream,
                       part.data_ptr<float>(), S, total, K, nullptr,
                       (unsigned short*)dx.data_ptr());
  }
  TORCH_CHECK(hipGetLastError() == hipSuccess, "lin_dgrad_small launch failed");
  return dx;
}

at::Tensor biasgelu_bwd_ew(at::Tensor dy, at::Tensor x, at::Tensor bias) {
  const int H = (int)bias.numel();
  check_bf16_2d(dy, "dy", H);
  auto dx = at::empty_like(dy);
  const long long total = dy.numel();
  const int R2 = (int)(total / H);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  int cblocks = (H / 8 + 255) / 256;
  hipLaunchKernelGGL(k_biasgelu_bwd_ew, dim3(cblocks, R2), dim3(256), 0, stream,
                     (const unsigned short*)dy.data_ptr(),
                     (const unsigned short*)x.data_ptr(),
                     (const unsigned short*)bias.data_ptr(),
                     (unsigned short*)dx.data_ptr(), total, H);
  TORCH_CHECK(hipGetLastError() == hipSuccess, "k_biasgelu_bwd_ew launch failed");
  return dx;
}

void colreduce_batch(std::vector<at::Tensor> parts,
                     std::vector<at::Tensor> d0s,
                     std::vector<c10::optional<at::Tensor>> d1s,
                     std::vector<c10::optional<at::Tensor>> d2s) {
  const int G = (int)parts.size();
  TORCH_CHECK(G > 0 && G <= CRB_MAX_G, "colreduce_batch: 1..16 problems");
  CrbArgs args{};
  args.G = G;
  long long blocks = 0;
  for (int g = 0; g < G; ++g) {
    const int NB = (int)parts[g].size(0);
    const int C = (int)(parts[g].numel() / NB);
    const int na = (int)d0s[g].numel();
    const int nb_ = d1s[g] ? (int)d1s[g]->numel() : 0;
    const int nc = d2s[g] ? (int)d2s[g]->numel() : 0;
    TORCH_CHECK(na + nb_ + nc == C, "colreduce_batch dest sizes");
    args.part[g] = (unsigned long long)parts[g].data_ptr();
    args.d0[g] = (unsigned long long)d0s[g].data_ptr();
    args.d1[g] = d1s[g] ? (unsigned long long)d1s[g]->data_ptr() : 0ull;
    args.d2[g] = d2s[g] ? (unsigned long long)d2s[g]->data_ptr() : 0ull;
    args.nb[g] = NB;
    args.c[g] = C;
    args.n0[g] = na;
    args.n1[g] = nb_;
    const int nch = (NB + 7) / 8;
    blocks += ((long long)C * nch + 255) / 256;
  }
  auto stream = c10::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(k_colreduce_batch, dim3((unsigned)blocks), dim3(256), 0,
                     stream, args);
  TORCH_CHECK(hipGetLastError() == hipSuccess, "k_colreduce_batch launch failed");
}

void embgrad_acc(at::Tensor dy, at::Tensor ids, at::Tensor accum_slice, int64_t H) {
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous() && dy.scalar_type() == at::kBFloat16,
              "dy must be contiguous bf16");
  TORCH_CHECK(ids.is_cuda() && ids.is_contiguous() && ids.scalar_type() == at::kLong,
              "ids must be contiguous int64");
  TORCH_CHECK(accum_slice.scalar_type() == at::kFloat && accum_slice.is_contiguous());
  TORCH_CHECK(H % 4 == 0 && dy.numel() == ids.numel() * H, "shape mismatch");
  const long long R = ids.numel();
  auto stream = c10::hip::getCurrentHIPStream().stream();
  long long total = R * (H / 4);
  int blocks = (int)std::min<long long>((total + 255) / 256, 2048);
  hipLaunchKernelGGL(k_embgrad_acc, dim3(blocks), dim3(256), 0, stream,
                     (const unsigned short*)dy.data_ptr(),
                     (const long long*)ids.data_ptr<int64_t>(),
                     accum_slice.data_ptr<float>(), R, (int)H);
  TORCH_CHECK(hipGetLastError() == hipSuccess, "k_embgrad_acc launch failed");
}

}  // namespace

void register_blas_acc(pybind11::module_& mod);
void register_lt_gemm(pybind11::module_& mod);
void register_grouped_wgrad(pybind11::module_& mod);
void register_ffn_mfma(pybind11::module_& mod);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, mod) {
  register_blas_acc(mod);
  register_lt_gemm(mod);
  register_grouped_wgrad(mod);
  register_ffn_mfma(mod);
  mod.def("accumulate", &accumulate, "accum += grad (fp32 upcast); grad = 0");
  mod.def("sqnorm", &sqnorm, "out[0] = sum(accum^2)");
  mod.def("fused_apply", &fused_apply,
          "normalize + clip + AdamWeightDecay + bf16 write-back + zero accum",
          pybind11::arg("accum"), pybind11::arg("m"), pybind11::arg("v"),
          pybind11::arg("master"), pybind11::arg("model"),
          pybind11::arg("has_model"), pybind11::arg("lr_dev"),
          pybind11::arg("sqnorm_ws"), pybind11::arg("decay_boundary"),
          pybind11::arg("inv_k"), pybind11::arg("clip"),
          pybind11::arg("weight_decay"), pybind11::arg("beta1"),
          pybind11::arg("beta2"), pybind11::arg("eps"),
          pybind11::arg("skip_norm") = false);
  mod.def("addln_fwd", &addln_fwd, "fused residual+bias+LayerNorm forward");
  mod.def("addln_bwd", &addln_bwd, "fused LayerNorm backward -> dh + fp32 partials");
  mod.def("biasgelu_fwd", &biasgelu_fwd, "y = gelu_tanh(x + bias)");
  mod.def("biasgelu_bwd", &biasgelu_bwd, "dx + fp32 dbias partials");
  mod.def("colreduce_acc", &colreduce_acc,
          "reduce partials over blocks, ADD into flat fp32 accum slices");
  mod.def("emb3_fwd", &emb3_fwd, "fused word+position(+type) embedding gather-sum");
  mod.def("lin_fwd_small", &lin_fwd_small,
          "small-GEMM MFMA forward (64x64 tiles, split-K >= 1024)");
  mod.def("lin_dgrad_small", &lin_dgrad_small,
          "small-GEMM MFMA dgrad (W^T LDS transpose staging)");
  mod.def("biasgelu_bwd_ew", &biasgelu_bwd_ew,
          "elementwise gelu backward (bias grad delegated to wgrad colsum)");
  mod.def("colreduce_batch", &colreduce_batch,
          "one launch reducing every pending LN/GELU partial slab into accum");
  mod.def("embgrad_acc", &embgrad_acc,
          "scatter-add embedding grads into the flat fp32 accum slice");
  mod.def("attn_fwd", &attn_fwd, "fused MFMA attention fwd (packed qkv)",
          pybind11::arg("qkv"), pybind11::arg("nh"),
          pybind11::arg("mask") = c10::nullopt,
          pybind11::arg("seed") = c10::nullopt, pybind11::arg("p_drop") = 0.0);
  mod.def("cls_head_fwd", &cls_head_fwd, "tanh+classifier+CE forward");
  mod.def("cls_head_bwd", &cls_head_bwd, "fused head backward -> d(pre-tanh)");
  mod.def("wgrad_mfma", &wgrad_mfma,
          "batched MFMA wgrad: accum_g += dy_g^T @ x_g over a tile table");
  mod.def("attn_bwd", &attn_bwd, "fused MFMA attention bwd -> packed dqkv",
          pybind11::arg("qkv"), pybind11::arg("out"), pybind11::arg("dout"),
          pybind11::arg("lse"), pybind11::arg("nh"),
          pybind11::arg("mask") = c10::nullopt,
          pybind11::arg("seed") = c10::nullopt, pybind11::arg("p_drop") = 0.0);
}
