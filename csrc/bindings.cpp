// pybind11 bindings for parallel_cnn_amd._C
//
// CPU ops take at::Tensor directly.  GPU launchers live in
// csrc/hip/lenet_kernels.hip (compiled by hipcc for gfx950, linked in as
// objects); this translation unit deliberately includes no HIP headers —
// streams cross the boundary as opaque pointers obtained from
// torch.cuda.current_stream().cuda_stream on the Python side.

#include <torch/extension.h>

#include "lenet_dims.h"

namespace pcnn {
void cpu_forward(at::Tensor x, at::Tensor params, at::Tensor a1, at::Tensor a2,
                 at::Tensor y);
double cpu_backward(at::Tensor x, at::Tensor params, at::Tensor a1,
                    at::Tensor a2, at::Tensor y, at::Tensor labels,
                    at::Tensor dz, at::Tensor dz2, at::Tensor dz1,
                    at::Tensor grads);
void cpu_update(at::Tensor params, at::Tensor grads, double dt, double scale);
}  // namespace pcnn

extern "C" {
int pcnn_launch_fwdbwd(const void* x, const float* params, void* a1, void* a2,
                       float* y, float* dz, float* dz2, float* dz1,
                       const int* labels, float* loss_accum, int* correct,
                       int B, int act_is_bf16, int mode, void* stream);
int pcnn_launch_wgrad(const void* x, const void* a1, const void* a2,
                      const float* dz, const float* dz2, const float* dz1,
                      float* grads, int B, int act_is_bf16, int chunk_imgs,
                      void* stream);
int pcnn_launch_wgrad_ex(const void* x, const void* a1, const void* a2,
                         const float* dz, const float* dz2, const float* dz1,
                         float* grads, int B, int act_is_bf16, int chunk_imgs,
                         int roles, void* stream);
int pcnn_launch_update(float* params, float* grads, float step, void* stream);
const char* pcnn_hip_error_string(int err);
}

namespace {

void check_hip(int err, const char* what) {
  TORCH_CHECK(err == 0, "HIP error in ", what, ": ",
              pcnn_hip_error_string(err));
}

int act_flag(const at::Tensor& t) {
  if (t.scalar_type() == at::kBFloat16) return 1;
  if (t.scalar_type() == at::kHalf) return 2;
  TORCH_CHECK(t.scalar_type() == at::kFloat,
              "activation tensors must be bf16, fp16 or fp32");
  return 0;
}

// mode: 0 = train, 1 = eval (argmax + correct count), 2 = infer
void hip_fwdbwd(at::Tensor x, at::Tensor params, at::Tensor a1, at::Tensor a2,
                at::Tensor y, at::Tensor dz, at::Tensor dz2, at::Tensor dz1,
                at::Tensor labels, at::Tensor loss_accum,
                at::Tensor correct_accum, int64_t B, int64_t mode,
                int64_t stream) {
  TORCH_CHECK(x.is_cuda() && params.is_cuda(), "expected device tensors");
  TORCH_CHECK(labels.scalar_type() == at::kInt, "labels must be int32");
  int f = act_flag(x);
  check_hip(pcnn_launch_fwdbwd(
                x.data_ptr(), params.data_ptr<float>(), a1.data_ptr(),
                a2.data_ptr(), y.numel() ? y.data_ptr<float>() : nullptr,
                dz.numel() ? dz.data_ptr<float>() : nullptr,
                dz2.numel() ? dz2.data_ptr<float>() : nullptr,
                dz1.numel() ? dz1.data_ptr<float>() : nullptr,
                labels.data_ptr<int>(),
                loss_accum.numel() ? loss_accum.data_ptr<float>() : nullptr,
                correct_accum.numel() ? correct_accum.data_ptr<int>() : nullptr,
                (int)B, f, (int)mode, (void*)stream),
            "fwdbwd");
}

void hip_wgrad(at::Tensor x, at::Tensor a1, at::Tensor a2, at::Tensor dz,
               at::Tensor dz2, at::Tensor dz1, at::Tensor grads, int64_t B,
               int64_t chunk_imgs, int64_t stream) {
  int f = act_flag(x);
  check_hip(pcnn_launch_wgrad(x.data_ptr(), a1.data_ptr(), a2.data_ptr(),
                              dz.data_ptr<float>(), dz2.data_ptr<float>(),
                              dz1.data_ptr<float>(), grads.data_ptr<float>(),
                              (int)B, f, (int)chunk_imgs, (void*)stream),
            "wgrad");
}

// roles bitmask (1=conv1, 2=pool, 4=fc) — ablation/diagnostics only.
void hip_wgrad_roles(at::Tensor x, at::Tensor a1, at::Tensor a2,
                     at::Tensor dz, at::Tensor dz2, at::Tensor dz1,
                     at::Tensor grads, int64_t B, int64_t chunk_imgs,
                     int64_t roles, int64_t stream) {
  int f = act_flag(x);
  check_hip(pcnn_launch_wgrad_ex(x.data_ptr(), a1.data_ptr(), a2.data_ptr(),
                                 dz.data_ptr<float>(), dz2.data_ptr<float>(),
                                 dz1.data_ptr<float>(), grads.data_ptr<float>(),
                                 (int)B, f, (int)chunk_imgs, (int)roles,
                                 (void*)stream),
            "wgrad_roles");
}

void hip_update(at::Tensor params, at::Tensor grads, double step,
                int64_t stream) {
  check_hip(pcnn_launch_update(params.data_ptr<float>(),
                               grads.data_ptr<float>(), (float)step,
                               (void*)stream),
            "update");
}

// Single-GPU fused training loop: enqueues `steps` full training steps
// (fwd+bwd-data -> wgrad -> update) from C++, cycling a device-resident
// batch pool.  Asynchronous — caller syncs the stream.  The distributed
// path keeps the per-step Python loop (the all-reduce sits between wgrad
// and update there).
void hip_train_steps(at::Tensor x_pool, at::Tensor labels_pool,
                     at::Tensor params, at::Tensor grads, at::Tensor a1,
                     at::Tensor a2, at::Tensor y, at::Tensor dz,
                     at::Tensor dz2, at::Tensor dz1, at::Tensor loss_accum,
                     int64_t B, int64_t steps, int64_t chunk_imgs,
                     double step_scale, int64_t stream) {
  TORCH_CHECK(x_pool.is_cuda() && x_pool.dim() == 2, "x_pool [P*B, 784]");
  TORCH_CHECK(labels_pool.scalar_type() == at::kInt, "labels must be int32");
  const int64_t pool_rows = x_pool.size(0);
  TORCH_CHECK(pool_rows % B == 0, "pool rows must be a multiple of B");
  const int64_t P = pool_rows / B;
  const int f = act_flag(x_pool);
  const size_t esz = f ? 2 : 4;  // bf16/fp16 vs fp32
  const char* xp = (const char*)x_pool.data_ptr();
  const int* lp = labels_pool.data_ptr<int>();
  float* pp = params.data_ptr<float>();
  float* gp = grads.data_ptr<float>();
  void* s = (void*)stream;
  for (int64_t st = 0; st < steps; ++st) {
    const int64_t i = st % P;
    const void* xb = xp + (size_t)i * B * pcnn::IN_PIX * esz;
    const int* lb = lp + i * B;
    check_hip(pcnn_launch_fwdbwd(xb, pp, a1.data_ptr(), a2.data_ptr(),
                                 y.data_ptr<float>(), dz.data_ptr<float>(),
                                 dz2.data_ptr<float>(), dz1.data_ptr<float>(),
                                 lb, loss_accum.data_ptr<float>(), nullptr,
                                 (int)B, f, 0, s),
              "train_steps/fwdbwd");
    check_hip(pcnn_launch_wgrad(xb, a1.data_ptr(), a2.data_ptr(),
                                dz.data_ptr<float>(), dz2.data_ptr<float>(),
                                dz1.data_ptr<float>(), gp, (int)B, f,
                                (int)chunk_imgs, s),
              "train_steps/wgrad");
    check_hip(pcnn_launch_update(pp, gp, (float)step_scale, s),
              "train_steps/update");
  }
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "parallel_cnn_amd native ops (CPU reference + gfx950 HIP kernels)";
  m.def("cpu_forward", &pcnn::cpu_forward);
  m.def("cpu_backward", &pcnn::cpu_backward);
  m.def("cpu_update", &pcnn::cpu_update);
  m.def("hip_fwdbwd", &hip_fwdbwd);
  m.def("hip_wgrad", &hip_wgrad);
  m.def("hip_wgrad_roles", &hip_wgrad_roles);
  m.def("hip_update", &hip_update);
  m.def("hip_train_steps", &hip_train_steps);
  m.attr("N_PARAMS") = pcnn::N_PARAMS;
  m.attr("OFF_C1W") = pcnn::OFF_C1W;
  m.attr("OFF_C1B") = pcnn::OFF_C1B;
  m.attr("OFF_S1W") = pcnn::OFF_S1W;
  m.attr("OFF_S1B") = pcnn::OFF_S1B;
  m.attr("OFF_FW") = pcnn::OFF_FW;
  m.attr("OFF_FB") = pcnn::OFF_FB;
  m.attr("REF_DT") = pcnn::REF_DT;
  m.attr("REF_THRESHOLD") = pcnn::REF_THRESHOLD;
  m.attr("HAS_HIP_KERNELS") = true;
}
