// PyTorch bindings for the MI355X gradient-accumulation kernels.
// Native HIP path only (no CUDA shim, no hipify): launches on the current
// c10 HIP stream so torch.cuda.graphs capture works.

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
                 double weight_decay, double beta1, double beta2, double eps) {
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

  if (clip > 0.0) {
    hipMemsetAsync(sqnorm_ws.data_ptr<float>(), 0, sizeof(float), stream);
    hipLaunchKernelGGL(k_sqnorm, grid_for(n4), dim3(GA_THREADS), 0, stream,
                       (const float4*)accum.data_ptr<float>(), n4,
                       sqnorm_ws.data_ptr<float>());
  }
  if (has_model) {
    TORCH_CHECK(model.scalar_type() == at::kBFloat16 && model.numel() == n,
                "model buffer must be flat bf16 of same length");
    hipLaunchKernelGGL(k_apply_bf16, grid_for(n4), dim3(GA_THREADS), 0, stream,
                       (float4*)accum.data_ptr<float>(), (float4*)m.data_ptr<float>(),
                       (float4*)v.data_ptr<float>(), (float4*)master.data_ptr<float>(),
                       (ushort4*)model.data_ptr(),
                       lr_dev.data_ptr<float>(), sqnorm_ws.data_ptr<float>(),
                       n4, decay_boundary / 4,
                       (float)inv_k, (float)clip, (float)weight_decay,
                       (float)beta1, (float)beta2, (float)eps);
  } else {
    hipLaunchKernelGGL(k_apply_f32, grid_for(n4), dim3(GA_THREADS), 0, stream,
                       (float4*)accum.data_ptr<float>(), (float4*)m.data_ptr<float>(),
                       (float4*)v.data_ptr<float>(), (float4*)master.data_ptr<float>(),
                       lr_dev.data_ptr<float>(), sqnorm_ws.data_ptr<float>(),
                       n4, decay_boundary / 4,
                       (float)inv_k, (float)clip, (float)weight_decay,
                       (float)beta1, (float)beta2, (float)eps);
  }
  TORCH_CHECK(hipGetLastError() == hipSuccess, "k_apply launch failed");
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, mod) {
  mod.def("accumulate", &accumulate, "accum += grad (fp32 upcast); grad = 0");
  mod.def("sqnorm", &sqnorm, "out[0] = sum(accum^2)");
  mod.def("fused_apply", &fused_apply,
          "normalize + clip + AdamWeightDecay + bf16 write-back + zero accum");
}
