// pcnn_train — native single-GPU CLI trainer (no Python, no torch).
//
// The framework equivalent of the reference's native driver
// (Sequential/Main.cpp & CUDA/main.cu: main -> loaddata -> learn -> test),
// batched and MI355X-native: one training step = the framework's 3 fused
// gfx950 kernels (csrc/hip/lenet_kernels.hip), bf16 activations, device-
// resident data, synced timing.  Same stdout shape as the reference
// ("Learning", per-epoch "error: %e", "Error Rate: %.2f%%").
//
// Build (setup.py does this automatically):
//   hipcc --offload-arch=gfx950 -O3 tools/pcnn_train.cpp \
//         csrc/hip/lenet_kernels.o -o tools/pcnn_train
//
// Run:  ./tools/pcnn_train [--epochs 1] [--batch-size 64] [--dt 0.1]
//       [--threshold 1e-2] [--train-count 60000] [--test-count 10000]
//       [--data synthetic|mnist] [--data-dir data] [--seed 0]
//       [--grad-reduction mean|sum] [--ckpt-save w.bin] [--ckpt-load w.bin]

#include <hip/hip_runtime.h>

#include <chrono>
#include <cmath>
#include <cstdint>
#include <cstdio>
#include <cstring>
#include <fstream>
#include <random>
#include <string>
#include <vector>

#include "../csrc/lenet_dims.h"

using namespace pcnn;

extern "C" {
int pcnn_launch_fwdbwd(const void* x, const float* params, void* a1, void* a2,
                       float* y, float* dz, float* dz2, float* dz1,
                       const int* labels, float* loss_accum, int* correct,
                       int B, int act_is_bf16, int mode, void* stream);
int pcnn_launch_wgrad(const void* x, const void* a1, const void* a2,
                      const float* dz, const float* dz2, const float* dz1,
                      float* grads, int B, int act_is_bf16, int chunk_imgs,
                      void* stream);
int pcnn_launch_update(float* params, float* grads, float step, void* stream);
const char* pcnn_hip_error_string(int err);
}

#define CHECK(x)                                                        \
  do {                                                                  \
    hipError_t e_ = (hipError_t)(x);                                    \
    if (e_ != hipSuccess) {                                             \
      fprintf(stderr, "HIP error %s at %s:%d\n", hipGetErrorString(e_), \
              __FILE__, __LINE__);                                      \
      return 1;                                                         \
    }                                                                   \
  } while (0)

static uint16_t f2bf(float f) {
  uint32_t u;
  memcpy(&u, &f, 4);
  uint32_t lsb = (u >> 16) & 1;            // round-to-nearest-even
  u += 0x7fffu + lsb;
  return (uint16_t)(u >> 16);
}

struct Args {
  int epochs = 1;
  int batch = 64;
  float dt = REF_DT;
  float threshold = REF_THRESHOLD;
  int train_count = 60000;
  int test_count = 10000;
  std::string data = "synthetic";
  std::string data_dir = "data";
  std::string ckpt_save, ckpt_load;
  std::string reduction = "mean";
  unsigned seed = 0;
};

static bool load_idx_images(const std::string& p, std::vector<float>& out,
                            int& n) {
  std::ifstream f(p, std::ios::binary);
  if (!f) return false;
  auto rd32 = [&]() {
    uint8_t b[4];
    f.read((char*)b, 4);
    return (int)((b[0] << 24) | (b[1] << 16) | (b[2] << 8) | b[3]);
  };
  if (rd32() != 2051) return false;
  n = rd32();
  int rows = rd32(), cols = rd32();
  if (rows != IN_H || cols != IN_W) return false;
  std::vector<uint8_t> raw((size_t)n * IN_PIX);
  f.read((char*)raw.data(), raw.size());
  if (!f) return false;
  out.resize(raw.size());
  for (size_t i = 0; i < raw.size(); ++i) out[i] = raw[i] / 255.0f;
  return true;
}

static bool load_idx_labels(const std::string& p, std::vector<int>& out,
                            int& n) {
  std::ifstream f(p, std::ios::binary);
  if (!f) return false;
  auto rd32 = [&]() {
    uint8_t b[4];
    f.read((char*)b, 4);
    return (int)((b[0] << 24) | (b[1] << 16) | (b[2] << 8) | b[3]);
  };
  if (rd32() != 2049) return false;
  n = rd32();
  std::vector<uint8_t> raw(n);
  f.read((char*)raw.data(), n);
  if (!f) return false;
  out.assign(raw.begin(), raw.end());
  return true;
}

static void synthetic(int n, unsigned seed, std::vector<float>& x,
                      std::vector<int>& y) {
  std::mt19937 g(seed);
  std::uniform_real_distribution<float> du(0.f, 1.f);
  std::uniform_int_distribution<int> dl(0, 9);
  x.resize((size_t)n * IN_PIX);
  y.resize(n);
  for (auto& v : x) v = du(g);
  for (int i = 0; i < n; ++i) {
    y[i] = dl(g);
    // label-dependent bright band (learnable signal)
    for (int r = 2 + y[i] * 2; r < 2 + y[i] * 2 + 3 && r < IN_H; ++r)
      for (int c = 0; c < IN_W; ++c) {
        float& p = x[(size_t)i * IN_PIX + r * IN_W + c];
        p = fminf(1.f, p * 0.3f + 0.7f);
      }
  }
}

int main(int argc, char** argv) {
  Args a;
  for (int i = 1; i < argc; ++i) {
    std::string k = argv[i];
    auto next = [&]() { return std::string(argv[++i]); };
    if (k == "--epochs") a.epochs = atoi(next().c_str());
    else if (k == "--batch-size") a.batch = atoi(next().c_str());
    else if (k == "--dt") a.dt = atof(next().c_str());
    else if (k == "--threshold") a.threshold = atof(next().c_str());
    else if (k == "--train-count") a.train_count = atoi(next().c_str());
    else if (k == "--test-count") a.test_count = atoi(next().c_str());
    else if (k == "--data") a.data = next();
    else if (k == "--data-dir") a.data_dir = next();
    else if (k == "--ckpt-save") a.ckpt_save = next();
    else if (k == "--ckpt-load") a.ckpt_load = next();
    else if (k == "--grad-reduction") a.reduction = next();
    else if (k == "--seed") a.seed = (unsigned)atoi(next().c_str());
    else {
      fprintf(stderr, "unknown flag %s\n", k.c_str());
      return 2;
    }
  }

  // ---- loaddata ----
  std::vector<float> xtr, xte;
  std::vector<int> ytr, yte;
  if (a.data == "mnist") {
    int n1, n2, n3, n4;
    if (!load_idx_images(a.data_dir + "/train-images.idx3-ubyte", xtr, n1) ||
        !load_idx_labels(a.data_dir + "/train-labels.idx1-ubyte", ytr, n2) ||
        !load_idx_images(a.data_dir + "/t10k-images.idx3-ubyte", xte, n3) ||
        !load_idx_labels(a.data_dir + "/t10k-labels.idx1-ubyte", yte, n4) ||
        n1 != n2 || n3 != n4) {
      fprintf(stderr, "failed to load MNIST from %s\n", a.data_dir.c_str());
      return 1;
    }
    a.train_count = n1;
    a.test_count = n3;
  } else {
    synthetic(a.train_count, a.seed, xtr, ytr);
    synthetic(a.test_count, a.seed + 1, xte, yte);
  }
  a.train_count = (a.train_count / a.batch) * a.batch;

  // ---- parameters: reference init (0.5 - rand in [0,1)) ----
  std::vector<float> params_h(N_PARAMS);
  {
    std::mt19937 g(a.seed + 1234);
    std::uniform_real_distribution<float> du(0.f, 1.f);
    for (auto& v : params_h) v = 0.5f - du(g);
  }
  if (!a.ckpt_load.empty()) {
    std::ifstream f(a.ckpt_load, std::ios::binary);
    if (!f.read((char*)params_h.data(), N_PARAMS * 4)) {
      fprintf(stderr, "bad checkpoint %s\n", a.ckpt_load.c_str());
      return 1;
    }
  }

  // ---- device buffers ----
  const int B = a.batch;
  auto to_bf16 = [](const std::vector<float>& v) {
    std::vector<uint16_t> o(v.size());
    for (size_t i = 0; i < v.size(); ++i) o[i] = f2bf(v[i]);
    return o;
  };
  auto xtr_bf = to_bf16(xtr), xte_bf = to_bf16(xte);
  void *d_xtr, *d_xte;
  int *d_ytr, *d_yte;
  float *d_params, *d_grads, *d_y, *d_dz, *d_dz2, *d_dz1, *d_loss;
  void *d_a1, *d_a2;
  int* d_correct;
  CHECK(hipMalloc(&d_xtr, xtr_bf.size() * 2));
  CHECK(hipMalloc(&d_xte, xte_bf.size() * 2));
  CHECK(hipMalloc(&d_ytr, ytr.size() * 4));
  CHECK(hipMalloc(&d_yte, yte.size() * 4));
  CHECK(hipMalloc(&d_params, N_PARAMS * 4));
  CHECK(hipMalloc(&d_grads, N_PARAMS * 4));
  CHECK(hipMalloc(&d_a1, (size_t)B * C1_OUT * 2));
  CHECK(hipMalloc(&d_a2, (size_t)B * S1_OUT * 2));
  CHECK(hipMalloc(&d_y, (size_t)B * FC_OUT * 4));
  CHECK(hipMalloc(&d_dz, (size_t)B * FC_OUT * 4));
  CHECK(hipMalloc(&d_dz2, (size_t)B * S1_OUT * 4));
  CHECK(hipMalloc(&d_dz1, (size_t)B * C1_OUT * 4));
  CHECK(hipMalloc(&d_loss, 4));
  CHECK(hipMalloc(&d_correct, 4));
  CHECK(hipMemcpy(d_xtr, xtr_bf.data(), xtr_bf.size() * 2,
                  hipMemcpyHostToDevice));
  CHECK(hipMemcpy(d_xte, xte_bf.data(), xte_bf.size() * 2,
                  hipMemcpyHostToDevice));
  CHECK(hipMemcpy(d_ytr, ytr.data(), ytr.size() * 4, hipMemcpyHostToDevice));
  CHECK(hipMemcpy(d_yte, yte.data(), yte.size() * 4, hipMemcpyHostToDevice));
  CHECK(hipMemcpy(d_params, params_h.data(), N_PARAMS * 4,
                  hipMemcpyHostToDevice));
  CHECK(hipMemset(d_grads, 0, N_PARAMS * 4));
  CHECK(hipMemset(d_loss, 0, 4));

  const float scale = a.reduction == "sum" ? 1.0f : 1.0f / (float)B;
  printf("Learning\n");
  int epochs_done = 0;
  auto t0 = std::chrono::steady_clock::now();
  for (int ep = 0; ep < a.epochs; ++ep) {
    for (int s = 0; s < a.train_count; s += B) {
      const void* xb = (const char*)d_xtr + (size_t)s * IN_PIX * 2;
      const int* yb = d_ytr + s;
      CHECK(pcnn_launch_fwdbwd(xb, d_params, d_a1, d_a2, d_y, d_dz, d_dz2,
                               d_dz1, yb, d_loss, d_correct, B, 1, 0,
                               nullptr));
      CHECK(pcnn_launch_wgrad(xb, d_a1, d_a2, d_dz, d_dz2, d_dz1, d_grads, B,
                              1, 0, nullptr));
      CHECK(pcnn_launch_update(d_params, d_grads, a.dt * scale, nullptr));
    }
    float loss_sum = 0.f;
    CHECK(hipMemcpy(&loss_sum, d_loss, 4, hipMemcpyDeviceToHost));
    CHECK(hipMemset(d_loss, 0, 4));
    CHECK(hipDeviceSynchronize());
    double secs = std::chrono::duration<double>(
                      std::chrono::steady_clock::now() - t0).count();
    printf("error: %e, time_on_gpu: %f\n", loss_sum / a.train_count, secs);
    ++epochs_done;
    if (loss_sum / a.train_count < a.threshold) break;  // reference early stop
  }
  CHECK(hipDeviceSynchronize());
  double total = std::chrono::duration<double>(
                     std::chrono::steady_clock::now() - t0).count();
  printf("\n Time - %f ms\n", total * 1e3);
  printf("epochs run: %d (early stop at error < %g)\n", epochs_done,
         a.threshold);
  printf("images/sec: %.0f\n",
         (double)epochs_done * a.train_count / total);

  if (!a.ckpt_save.empty()) {
    CHECK(hipMemcpy(params_h.data(), d_params, N_PARAMS * 4,
                    hipMemcpyDeviceToHost));
    std::ofstream f(a.ckpt_save, std::ios::binary);
    f.write((const char*)params_h.data(), N_PARAMS * 4);
    printf("saved checkpoint: %s\n", a.ckpt_save.c_str());
  }

  // ---- test ----
  CHECK(hipMemset(d_correct, 0, 4));
  for (int s = 0; s < a.test_count; s += B) {
    const int n = std::min(B, a.test_count - s);
    CHECK(pcnn_launch_fwdbwd((const char*)d_xte + (size_t)s * IN_PIX * 2,
                             d_params, d_a1, d_a2, d_y, d_dz, d_dz2, d_dz1,
                             d_yte + s, d_loss, d_correct, n, 1, 1, nullptr));
  }
  int correct = 0;
  CHECK(hipMemcpy(&correct, d_correct, 4, hipMemcpyDeviceToHost));
  printf("Error Rate: %.2f%%\n",
         100.0 * (1.0 - correct / (double)a.test_count));
  return 0;
}
