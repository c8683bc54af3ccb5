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
                 at::Tensor y, int64_t pool_mode, int64_t loss_mode);
double cpu_backward(at::Tensor x, at::Tensor params, at::Tensor a1,
                    at::Tensor a2, at::Tensor y, at::Tensor labels,
                    at::Tensor dz, at::Tensor dz2, at::Tensor dz1,
                    at::Tensor grads, int64_t pool_mode, int64_t loss_mode);
void cpu_update(at::Tensor params, at::Tensor grads, double dt, double scale);
}  // namespace pcnn

extern "C" {
int pcnn_launch_fwdbwd(const void* x, const float* params, void* a1, void* a2,
                       float* y, float* dz, float* dz2, void* dz1,
                       const int* labels, float* loss_accum, int* correct,
                       int B, int act_is_bf16, int mode, void* stream);
int pcnn_launch_fwdbwd_ex(const void* x, const float* params, void* a1,
                          void* a2, float* y, float* dz, float* dz2,
                          void* dz1, const int* labels, float* loss_accum,
                          int* correct, int B, int act_is_bf16, int mode,
                          int pool_mode, int loss_mode, void* stream);
int pcnn_launch_fwdbwd_ex2(const void* x, const float* params, void* a1,
                           void* a2, float* y, float* dz, float* dz2,
                           void* dz1, const int* labels, float* loss_accum,
                           int* correct, int B, int act_is_bf16, int mode,
                           int pool_mode, int loss_mode, float* grads,
                           int wgrad_fuse, void* stream);
int pcnn_launch_wgrad(const void* x, const void* a1, const void* a2,
                      const float* dz, const float* dz2, const void* dz1,
                      float* grads, int B, int act_is_bf16, int chunk_imgs,
                      void* stream);
int pcnn_launch_wgrad_ex(const void* x, const void* a1, const void* a2,
                         const float* dz, const float* dz2, const void* dz1,
                         float* grads, int B, int act_is_bf16, int chunk_imgs,
                         int roles, void* stream);
int pcnn_launch_update(float* params, float* grads, float step, void* stream);
const char* pcnn_hip_error_string(int err);
// deep (general conv) kernels — csrc/hip/conv_kernels.hip
int pcnn_deep_im2col(const void* x, void* cols, int B, int H, int W, int Cin,
                     int K, int P, int KcP, int actf, void* stream);
int pcnn_deep_gemm(const void* A, const float* Bsrc, const float* bias,
                   void* C, long long M, int K, int N, int ldA, int ldC,
                   int b_kxn, int epilogue, int actf, void* stream);
int pcnn_deep_gemm_ex4(const void* A, const float* Bsrc, const void* Bpre,
                       const float* bias, void* C, long long M, int K, int N,
                       int ldA, int ldC, int b_kxn, int epilogue,
                       const void* imx, int XH, int XW, int XC, int XK,
                       int XP, const void* epi, const float* pw, void* pout,
                       int PK, float* c32, long long c32_cap, int actf,
                       void* stream);
int pcnn_deep_wgrad_gemm_ex2(const void* cols, const void* dpre, float* dW,
                             float* part, long long M, int KcP, int N,
                             int MS, const void* imx, int XH, int XW, int XC,
                             int XK, int XP, float* db, int actf,
                             void* stream);
int pcnn_deep_cast_wt(const float* W, void* out, void* outT, int R, int C,
                      void* stream);
int pcnn_deep_cast_all(const float* params, void* wbuf, int n_stages,
                       const int* R, const int* C, const int* K,
                       const int* Cin, const long long* w_off,
                       const long long* bf_off, const long long* bfT_off,
                       const long long* rot_off, const long long* p8_off,
                       void* stream);
int pcnn_deep_wgrad_multi(int n_stages, const void* const* a,
                          const void* const* dpre, float* const* dW,
                          float* const* db, const long long* M,
                          const int* KcP, const int* N, const int* MS,
                          const int* implicit, const int* XH, const int* XW,
                          const int* XC, const int* XK, const int* XP,
                          int actf, void* stream);
int pcnn_deep_update_cast(float* params, float* grads, long long n,
                          float step, void* wbuf, int n_stages, const int* R,
                          const int* C, const int* K, const int* Cin,
                          const long long* w_off, const long long* bf_off,
                          const long long* bfT_off, const long long* rot_off,
                          const long long* p8_off, void* stream);
int pcnn_deep_pad_channels(const void* x, void* x8, long long npix, int Cin,
                           int actf, void* stream);
int pcnn_deep_remap_dw8(float* dW8, float* dW, int KK, int Cin, int Cout,
                        void* stream);
int pcnn_deep_wgrad_gemm(const void* cols, const void* dpre, float* dW,
                         float* part, long long M, int KcP, int N, int MS,
                         int actf, void* stream);
int pcnn_deep_colsum(const void* dpre, float* part, float* db, long long M,
                     int N,
                     int slices, int actf, void* stream);
int pcnn_deep_col2im_sigbwd(const void* dcols, const void* pout, void* out,
                            int B, int H, int W, int Cin, int K, int P,
                            int KcP, int actf, void* stream);
int pcnn_deep_pool_fwd(const void* a, const float* pw, void* pout, int B,
                       int H, int W, int C, int K, int actf, void* stream);
int pcnn_deep_pool_bwd(const void* dppre, const void* a, const float* pw,
                       void* dapre, int B, int H, int W, int C, int K,
                       int actf, void* stream);
int pcnn_deep_pool_wgrad(const void* dppre, const void* a, float* dpw, int B,
                         int H, int W, int C, int K, int G, int actf,
                         void* stream);
int pcnn_deep_pool_wbwd(const void* dppre, const void* a, const float* pw,
                        void* dapre, float* dpw, int B, int H, int W, int C,
                        int K, int G, int actf, void* stream);
int pcnn_deep_fc_fwd2(const void* flat, const float* fw, const float* fb,
                      const int* labels, float* yg, float* dzg,
                      float* loss_accum, int* correct_accum, int B, int FCIN,
                      int NCLS, int mode, void* dflat, int actf,
                      void* stream);
int pcnn_deep_fc_bwd(const float* dzg, const void* flat, const float* fw,
                     void* dflat, int B, int FCIN, int NCLS, int actf,
                     void* stream);
int pcnn_deep_fc_wgrad(const float* dzg, const void* flat, float* gfw,
                       float* gfb, int B, int FCIN, int NCLS, int FS,
                       int actf, void* stream);
int pcnn_deep_update(float* params, float* grads, long long n, float step,
                     void* stream);
int pcnn_deep_mfma_selftest(const float* A, const float* Bm, float* D,
                            void* stream);
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
                int64_t stream, int64_t pool_mode, int64_t loss_mode,
                at::Tensor grads, int64_t wgrad_fuse) {
  TORCH_CHECK(x.is_cuda() && params.is_cuda(), "expected device tensors");
  TORCH_CHECK(labels.scalar_type() == at::kInt, "labels must be int32");
  TORCH_CHECK(!dz1.numel() || dz1.scalar_type() == x.scalar_type(),
              "dz1 must match the activation dtype");
  int f = act_flag(x);
  check_hip(pcnn_launch_fwdbwd_ex2(
                x.data_ptr(), params.data_ptr<float>(), a1.data_ptr(),
                a2.data_ptr(), y.numel() ? y.data_ptr<float>() : nullptr,
                dz.numel() ? dz.data_ptr<float>() : nullptr,
                dz2.numel() ? dz2.data_ptr<float>() : nullptr,
                dz1.numel() ? dz1.data_ptr() : nullptr,
                labels.data_ptr<int>(),
                loss_accum.numel() ? loss_accum.data_ptr<float>() : nullptr,
                correct_accum.numel() ? correct_accum.data_ptr<int>() : nullptr,
                (int)B, f, (int)mode, (int)pool_mode, (int)loss_mode,
                grads.numel() ? grads.data_ptr<float>() : nullptr,
                (int)wgrad_fuse, (void*)stream),
            "fwdbwd");
}

void hip_wgrad(at::Tensor x, at::Tensor a1, at::Tensor a2, at::Tensor dz,
               at::Tensor dz2, at::Tensor dz1, at::Tensor grads, int64_t B,
               int64_t chunk_imgs, int64_t stream) {
  int f = act_flag(x);
  TORCH_CHECK(dz1.scalar_type() == x.scalar_type(),
              "dz1 must match the activation dtype");
  check_hip(pcnn_launch_wgrad(x.data_ptr(), a1.data_ptr(), a2.data_ptr(),
                              dz.data_ptr<float>(), dz2.data_ptr<float>(),
                              dz1.data_ptr(), grads.data_ptr<float>(),
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
                                 dz1.data_ptr(), grads.data_ptr<float>(),
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
                     double step_scale, int64_t stream, int64_t pool_mode,
                     int64_t loss_mode, int64_t wgrad_fuse) {
  // wgrad_fuse=1: conv/pool grads accumulate inside the fwdbwd kernel and
  // kernel B covers only the fc role.  Measured slower at bs=64 (the LDS
  // atomic combine serializes) — default off.
  const int wroles = wgrad_fuse ? 4 : (pool_mode == 1 ? 5 : 7);
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
    check_hip(pcnn_launch_fwdbwd_ex2(xb, pp, a1.data_ptr(), a2.data_ptr(),
                                     y.data_ptr<float>(),
                                     dz.data_ptr<float>(),
                                     dz2.data_ptr<float>(),
                                     dz1.data_ptr(), lb,
                                     loss_accum.data_ptr<float>(), nullptr,
                                     (int)B, f, 0, (int)pool_mode,
                                     (int)loss_mode, gp, (int)wgrad_fuse, s),
              "train_steps/fwdbwd");
    check_hip(pcnn_launch_wgrad_ex(xb, a1.data_ptr(), a2.data_ptr(),
                                   dz.data_ptr<float>(), dz2.data_ptr<float>(),
                                   dz1.data_ptr(), gp, (int)B, f,
                                   (int)chunk_imgs, wroles, s),
              "train_steps/wgrad");
    check_hip(pcnn_launch_update(pp, gp, (float)step_scale, s),
              "train_steps/update");
  }
}

// ---------------------------------------------------------------------------
// DeepCNN (general conv path) wrappers
// ---------------------------------------------------------------------------
void deep_im2col(at::Tensor x, at::Tensor cols, int64_t B, int64_t H,
                 int64_t W, int64_t Cin, int64_t K, int64_t P, int64_t KcP,
                 int64_t stream) {
  check_hip(pcnn_deep_im2col(x.data_ptr(), cols.data_ptr(), (int)B, (int)H,
                             (int)W, (int)Cin, (int)K, (int)P, (int)KcP,
                             act_flag(x), (void*)stream),
            "deep_im2col");
}

void deep_gemm(at::Tensor A, at::Tensor Bsrc, at::Tensor bias, at::Tensor C,
               int64_t M, int64_t K, int64_t N, int64_t ldA, int64_t ldC,
               int64_t b_kxn, int64_t epilogue, int64_t stream,
               at::Tensor Bpre, at::Tensor imx, int64_t XH, int64_t XW,
               int64_t XC, int64_t XK, int64_t XP, at::Tensor epi,
               at::Tensor pw, at::Tensor pout, int64_t PK, at::Tensor c32) {
  check_hip(pcnn_deep_gemm_ex4(
                A.data_ptr(), Bsrc.data_ptr<float>(),
                Bpre.numel() ? Bpre.data_ptr() : nullptr,
                bias.numel() ? bias.data_ptr<float>() : nullptr,
                C.data_ptr(), M, (int)K, (int)N, (int)ldA, (int)ldC,
                (int)b_kxn, (int)epilogue,
                imx.numel() ? imx.data_ptr() : nullptr, (int)XH, (int)XW,
                (int)XC, (int)XK, (int)XP,
                epi.numel() ? epi.data_ptr() : nullptr,
                pw.numel() ? pw.data_ptr<float>() : nullptr,
                pout.numel() ? pout.data_ptr() : nullptr, (int)PK,
                c32.numel() ? c32.data_ptr<float>() : nullptr,
                (long long)c32.numel(), act_flag(A), (void*)stream),
            "deep_gemm");
}

void deep_cast_wt(at::Tensor W, at::Tensor out, at::Tensor outT, int64_t R,
                  int64_t C, int64_t stream) {
  check_hip(pcnn_deep_cast_wt(W.data_ptr<float>(), out.data_ptr(),
                              outT.data_ptr(), (int)R, (int)C,
                              (void*)stream),
            "deep_cast_wt");
}

void deep_cast_all(at::Tensor params, at::Tensor wbuf,
                   std::vector<int64_t> R, std::vector<int64_t> C,
                   std::vector<int64_t> K, std::vector<int64_t> Cin,
                   std::vector<int64_t> w_off, std::vector<int64_t> bf_off,
                   std::vector<int64_t> bfT_off,
                   std::vector<int64_t> rot_off,
                   std::vector<int64_t> p8_off, int64_t stream) {
  const size_t n = R.size();
  TORCH_CHECK(n >= 1 && n <= 8, "deep_cast_all: 1..8 stages");
  TORCH_CHECK(C.size() == n && K.size() == n && Cin.size() == n &&
                  w_off.size() == n && bf_off.size() == n &&
                  bfT_off.size() == n && rot_off.size() == n &&
                  (p8_off.empty() || p8_off.size() == n),
              "deep_cast_all: descriptor length mismatch");
  int Ri[8], Ci[8], Ki[8], Cini[8];
  long long wo[8], bo[8], bto[8], ro[8], po[8];
  for (size_t s = 0; s < n; ++s) {
    Ri[s] = (int)R[s];
    Ci[s] = (int)C[s];
    Ki[s] = (int)K[s];
    Cini[s] = (int)Cin[s];
    wo[s] = w_off[s];
    bo[s] = bf_off[s];
    bto[s] = bfT_off[s];
    ro[s] = rot_off[s];
    po[s] = p8_off.empty() ? -1 : p8_off[s];
  }
  check_hip(pcnn_deep_cast_all(params.data_ptr<float>(), wbuf.data_ptr(),
                               (int)n, Ri, Ci, Ki, Cini, wo, bo, bto, ro,
                               po, (void*)stream),
            "deep_cast_all");
}

void deep_update_cast(at::Tensor params, at::Tensor grads, double step,
                      at::Tensor wbuf, std::vector<int64_t> R,
                      std::vector<int64_t> C, std::vector<int64_t> K,
                      std::vector<int64_t> Cin, std::vector<int64_t> w_off,
                      std::vector<int64_t> bf_off,
                      std::vector<int64_t> bfT_off,
                      std::vector<int64_t> rot_off,
                      std::vector<int64_t> p8_off, int64_t stream) {
  const size_t n = R.size();
  TORCH_CHECK(n >= 1 && n <= 8, "deep_update_cast: 1..8 stages");
  int Ri[8], Ci[8], Ki[8], Cini[8];
  long long wo[8], bo[8], bto[8], ro[8], po[8];
  for (size_t s = 0; s < n; ++s) {
    Ri[s] = (int)R[s];
    Ci[s] = (int)C[s];
    Ki[s] = (int)K[s];
    Cini[s] = (int)Cin[s];
    wo[s] = w_off[s];
    bo[s] = bf_off[s];
    bto[s] = bfT_off[s];
    ro[s] = rot_off[s];
    po[s] = p8_off.empty() ? -1 : p8_off[s];
  }
  check_hip(pcnn_deep_update_cast(params.data_ptr<float>(),
                                  grads.data_ptr<float>(), params.numel(),
                                  (float)step, wbuf.data_ptr(), (int)n, Ri,
                                  Ci, Ki, Cini, wo, bo, bto, ro, po,
                                  (void*)stream),
            "deep_update_cast");
}

void deep_wgrad_multi(std::vector<at::Tensor> a,
                      std::vector<at::Tensor> dpre,
                      std::vector<at::Tensor> dW, std::vector<at::Tensor> db,
                      std::vector<int64_t> M, std::vector<int64_t> KcP,
                      std::vector<int64_t> N, std::vector<int64_t> MS,
                      std::vector<int64_t> implicit, std::vector<int64_t> XH,
                      std::vector<int64_t> XW, std::vector<int64_t> XC,
                      std::vector<int64_t> XK, std::vector<int64_t> XP,
                      int64_t stream) {
  const size_t n = a.size();
  TORCH_CHECK(n >= 1 && n <= 8, "deep_wgrad_multi: 1..8 stages");
  const void* ap[8];
  const void* dp[8];
  float* wp[8];
  float* bp[8];
  long long Mi[8];
  int Ki[8], Ni[8], MSi[8], Ii[8], XHi[8], XWi[8], XCi[8], XKi[8], XPi[8];
  for (size_t s = 0; s < n; ++s) {
    ap[s] = a[s].data_ptr();
    dp[s] = dpre[s].data_ptr();
    wp[s] = dW[s].data_ptr<float>();
    bp[s] = db[s].data_ptr<float>();
    Mi[s] = M[s];
    Ki[s] = (int)KcP[s];
    Ni[s] = (int)N[s];
    MSi[s] = (int)MS[s];
    Ii[s] = (int)implicit[s];
    XHi[s] = (int)XH[s];
    XWi[s] = (int)XW[s];
    XCi[s] = (int)XC[s];
    XKi[s] = (int)XK[s];
    XPi[s] = (int)XP[s];
  }
  check_hip(pcnn_deep_wgrad_multi((int)n, ap, dp, wp, bp, Mi, Ki, Ni, MSi,
                                  Ii, XHi, XWi, XCi, XKi, XPi,
                                  act_flag(dpre[0]), (void*)stream),
            "deep_wgrad_multi");
}

void deep_pad_channels(at::Tensor x, at::Tensor x8, int64_t npix,
                       int64_t Cin, int64_t stream) {
  check_hip(pcnn_deep_pad_channels(x.data_ptr(), x8.data_ptr(), npix,
                                   (int)Cin, act_flag(x), (void*)stream),
            "deep_pad_channels");
}

void deep_remap_dw8(at::Tensor dW8, at::Tensor dW, int64_t KK, int64_t Cin,
                    int64_t Cout, int64_t stream) {
  check_hip(pcnn_deep_remap_dw8(dW8.data_ptr<float>(), dW.data_ptr<float>(),
                                (int)KK, (int)Cin, (int)Cout,
                                (void*)stream),
            "deep_remap_dw8");
}

void deep_wgrad_gemm(at::Tensor cols, at::Tensor dpre, at::Tensor dW,
                     int64_t M, int64_t KcP, int64_t N, int64_t MS,
                     int64_t stream, at::Tensor imx, int64_t XH, int64_t XW,
                     int64_t XC, int64_t XK, int64_t XP, at::Tensor part,
                     at::Tensor db) {
  float* pp = nullptr;
  if (part.numel()) {
    TORCH_CHECK(part.numel() >= MS * KcP * N, "wgrad slab scratch too small");
    pp = part.data_ptr<float>();
  }
  check_hip(pcnn_deep_wgrad_gemm_ex2(
                cols.data_ptr(), dpre.data_ptr(), dW.data_ptr<float>(), pp, M,
                (int)KcP, (int)N, (int)MS,
                imx.numel() ? imx.data_ptr() : nullptr, (int)XH, (int)XW,
                (int)XC, (int)XK, (int)XP,
                db.numel() ? db.data_ptr<float>() : nullptr,
                act_flag(cols), (void*)stream),
            "deep_wgrad_gemm");
}

void deep_colsum(at::Tensor dpre, at::Tensor part, at::Tensor db, int64_t M,
                 int64_t N, int64_t slices, int64_t stream) {
  TORCH_CHECK(part.numel() >= slices * N, "colsum scratch too small");
  check_hip(pcnn_deep_colsum(dpre.data_ptr(), part.data_ptr<float>(),
                             db.data_ptr<float>(), M, (int)N, (int)slices,
                             act_flag(dpre), (void*)stream),
            "deep_colsum");
}

void deep_col2im_sigbwd(at::Tensor dcols, at::Tensor pout, at::Tensor out,
                        int64_t B, int64_t H, int64_t W, int64_t Cin,
                        int64_t K, int64_t P, int64_t KcP, int64_t stream) {
  check_hip(pcnn_deep_col2im_sigbwd(
                dcols.data_ptr(), pout.numel() ? pout.data_ptr() : nullptr,
                out.data_ptr(), (int)B, (int)H, (int)W, (int)Cin, (int)K,
                (int)P, (int)KcP, act_flag(dcols), (void*)stream),
            "deep_col2im_sigbwd");
}

void deep_pool_fwd(at::Tensor a, at::Tensor pw, at::Tensor pout, int64_t B,
                   int64_t H, int64_t W, int64_t C, int64_t K,
                   int64_t stream) {
  check_hip(pcnn_deep_pool_fwd(a.data_ptr(), pw.data_ptr<float>(),
                               pout.data_ptr(), (int)B, (int)H, (int)W,
                               (int)C, (int)K, act_flag(a), (void*)stream),
            "deep_pool_fwd");
}

void deep_pool_bwd(at::Tensor dppre, at::Tensor a, at::Tensor pw,
                   at::Tensor dapre, int64_t B, int64_t H, int64_t W,
                   int64_t C, int64_t K, int64_t stream) {
  check_hip(pcnn_deep_pool_bwd(dppre.data_ptr(), a.data_ptr(),
                               pw.data_ptr<float>(), dapre.data_ptr(),
                               (int)B, (int)H, (int)W, (int)C, (int)K,
                               act_flag(a), (void*)stream),
            "deep_pool_bwd");
}

void deep_pool_wgrad(at::Tensor dppre, at::Tensor a, at::Tensor dpw,
                     int64_t B, int64_t H, int64_t W, int64_t C, int64_t K,
                     int64_t G, int64_t stream) {
  check_hip(pcnn_deep_pool_wgrad(dppre.data_ptr(), a.data_ptr(),
                                 dpw.data_ptr<float>(), (int)B, (int)H,
                                 (int)W, (int)C, (int)K, (int)G,
                                 act_flag(a), (void*)stream),
            "deep_pool_wgrad");
}

void deep_pool_wbwd(at::Tensor dppre, at::Tensor a, at::Tensor pw,
                    at::Tensor dapre, at::Tensor dpw, int64_t B, int64_t H,
                    int64_t W, int64_t C, int64_t K, int64_t G,
                    int64_t stream) {
  check_hip(pcnn_deep_pool_wbwd(dppre.data_ptr(), a.data_ptr(),
                                pw.data_ptr<float>(), dapre.data_ptr(),
                                dpw.data_ptr<float>(), (int)B, (int)H,
                                (int)W, (int)C, (int)K, (int)G,
                                act_flag(a), (void*)stream),
            "deep_pool_wbwd");
}

void deep_fc_fwd(at::Tensor flat, at::Tensor fw, at::Tensor fb,
                 at::Tensor labels, at::Tensor y, at::Tensor dz,
                 at::Tensor loss_accum, at::Tensor correct_accum, int64_t B,
                 int64_t FCIN, int64_t NCLS, int64_t mode, int64_t stream,
                 at::Tensor dflat) {
  check_hip(pcnn_deep_fc_fwd2(
                flat.data_ptr(), fw.data_ptr<float>(), fb.data_ptr<float>(),
                labels.data_ptr<int>(),
                y.numel() ? y.data_ptr<float>() : nullptr,
                dz.numel() ? dz.data_ptr<float>() : nullptr,
                loss_accum.numel() ? loss_accum.data_ptr<float>() : nullptr,
                correct_accum.numel() ? correct_accum.data_ptr<int>()
                                      : nullptr,
                (int)B, (int)FCIN, (int)NCLS, (int)mode,
                dflat.numel() ? dflat.data_ptr() : nullptr, act_flag(flat),
                (void*)stream),
            "deep_fc_fwd");
}

void deep_fc_bwd(at::Tensor dz, at::Tensor flat, at::Tensor fw,
                 at::Tensor dflat, int64_t B, int64_t FCIN, int64_t NCLS,
                 int64_t stream) {
  check_hip(pcnn_deep_fc_bwd(dz.data_ptr<float>(), flat.data_ptr(),
                             fw.data_ptr<float>(), dflat.data_ptr(), (int)B,
                             (int)FCIN, (int)NCLS, act_flag(flat),
                             (void*)stream),
            "deep_fc_bwd");
}

void deep_fc_wgrad(at::Tensor dz, at::Tensor flat, at::Tensor gfw,
                   at::Tensor gfb, int64_t B, int64_t FCIN, int64_t NCLS,
                   int64_t FS, int64_t stream) {
  check_hip(pcnn_deep_fc_wgrad(dz.data_ptr<float>(), flat.data_ptr(),
                               gfw.data_ptr<float>(), gfb.data_ptr<float>(),
                               (int)B, (int)FCIN, (int)NCLS, (int)FS,
                               act_flag(flat), (void*)stream),
            "deep_fc_wgrad");
}

void deep_update(at::Tensor params, at::Tensor grads, double step,
                 int64_t stream) {
  check_hip(pcnn_deep_update(params.data_ptr<float>(),
                             grads.data_ptr<float>(), params.numel(),
                             (float)step, (void*)stream),
            "deep_update");
}

void deep_mfma_selftest(at::Tensor A, at::Tensor Bm, at::Tensor D,
                        int64_t stream) {
  check_hip(pcnn_deep_mfma_selftest(A.data_ptr<float>(),
                                    Bm.data_ptr<float>(),
                                    D.data_ptr<float>(), (void*)stream),
            "deep_mfma_selftest");
}


}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "parallel_cnn_amd native ops (CPU reference + gfx950 HIP kernels)";
  m.def("cpu_forward", &pcnn::cpu_forward, py::arg("x"), py::arg("params"),
        py::arg("a1"), py::arg("a2"), py::arg("y"), py::arg("pool_mode") = 0,
        py::arg("loss_mode") = 0);
  m.def("cpu_backward", &pcnn::cpu_backward, py::arg("x"), py::arg("params"),
        py::arg("a1"), py::arg("a2"), py::arg("y"), py::arg("labels"),
        py::arg("dz"), py::arg("dz2"), py::arg("dz1"), py::arg("grads"),
        py::arg("pool_mode") = 0, py::arg("loss_mode") = 0);
  m.def("cpu_update", &pcnn::cpu_update);
  m.def("hip_fwdbwd", &hip_fwdbwd, py::arg("x"), py::arg("params"),
        py::arg("a1"), py::arg("a2"), py::arg("y"), py::arg("dz"),
        py::arg("dz2"), py::arg("dz1"), py::arg("labels"),
        py::arg("loss_accum"), py::arg("correct_accum"), py::arg("B"),
        py::arg("mode"), py::arg("stream"), py::arg("pool_mode") = 0,
        py::arg("loss_mode") = 0, py::arg("grads") = at::empty({0}),
        py::arg("wgrad_fuse") = 0);
  m.def("hip_wgrad", &hip_wgrad);
  m.def("hip_wgrad_roles", &hip_wgrad_roles);
  m.def("hip_update", &hip_update);
  m.def("hip_train_steps", &hip_train_steps, py::arg("x_pool"),
        py::arg("labels_pool"), py::arg("params"), py::arg("grads"),
        py::arg("a1"), py::arg("a2"), py::arg("y"), py::arg("dz"),
        py::arg("dz2"), py::arg("dz1"), py::arg("loss_accum"), py::arg("B"),
        py::arg("steps"), py::arg("chunk_imgs"), py::arg("step_scale"),
        py::arg("stream"), py::arg("pool_mode") = 0,
        py::arg("loss_mode") = 0, py::arg("wgrad_fuse") = 0);
  m.def("deep_im2col", &deep_im2col);
  m.def("deep_gemm", &deep_gemm, py::arg("A"), py::arg("Bsrc"),
        py::arg("bias"), py::arg("C"), py::arg("M"), py::arg("K"),
        py::arg("N"), py::arg("ldA"), py::arg("ldC"), py::arg("b_kxn"),
        py::arg("epilogue"), py::arg("stream"),
        py::arg("Bpre") = at::empty({0}),
        py::arg("imx") = at::empty({0}), py::arg("XH") = 0,
        py::arg("XW") = 0, py::arg("XC") = 0, py::arg("XK") = 0,
        py::arg("XP") = 0, py::arg("epi") = at::empty({0}),
        py::arg("pw") = at::empty({0}), py::arg("pout") = at::empty({0}),
        py::arg("PK") = 0, py::arg("c32") = at::empty({0}));
  m.def("deep_cast_wt", &deep_cast_wt);
  m.def("deep_cast_all", &deep_cast_all);
  m.def("deep_pad_channels", &deep_pad_channels);
  m.def("deep_update_cast", &deep_update_cast);
  m.def("deep_wgrad_multi", &deep_wgrad_multi);
  m.def("deep_remap_dw8", &deep_remap_dw8);
  m.def("deep_wgrad_gemm", &deep_wgrad_gemm, py::arg("cols"),
        py::arg("dpre"), py::arg("dW"), py::arg("M"), py::arg("KcP"),
        py::arg("N"), py::arg("MS"), py::arg("stream"),
        py::arg("imx") = at::empty({0}), py::arg("XH") = 0,
        py::arg("XW") = 0, py::arg("XC") = 0, py::arg("XK") = 0,
        py::arg("XP") = 0, py::arg("part") = at::empty({0}),
        py::arg("db") = at::empty({0}));
  m.def("deep_colsum", &deep_colsum);
  m.def("deep_col2im_sigbwd", &deep_col2im_sigbwd);
  m.def("deep_pool_fwd", &deep_pool_fwd);
  m.def("deep_pool_bwd", &deep_pool_bwd);
  m.def("deep_pool_wgrad", &deep_pool_wgrad);
  m.def("deep_pool_wbwd", &deep_pool_wbwd);
  m.def("deep_fc_fwd", &deep_fc_fwd, py::arg("flat"), py::arg("fw"),
        py::arg("fb"), py::arg("labels"), py::arg("y"), py::arg("dz"),
        py::arg("loss_accum"), py::arg("correct_accum"), py::arg("B"),
        py::arg("FCIN"), py::arg("NCLS"), py::arg("mode"),
        py::arg("stream"), py::arg("dflat") = at::empty({0}));
  m.def("deep_fc_bwd", &deep_fc_bwd);
  m.def("deep_fc_wgrad", &deep_fc_wgrad);
  m.def("deep_update", &deep_update);
  m.def("deep_mfma_selftest", &deep_mfma_selftest);
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
