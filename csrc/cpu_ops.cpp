// CPU reference ops for the LeNet-5-class network (fp32, batched).
//
// This is the correctness oracle of the framework and the framework's own CPU
// execution path (the capability the reference delivers with its Sequential/
// and Openmp/ variants).  Math semantics follow SURVEY.md §0.1:
//   * sigmoid activation after conv, pool and fc (Sequential/layer.h:81-83)
//   * loss gradient: d_z = onehot(label) - y, used directly as the fc
//     pre-activation gradient (no sigmoid derivative at the output layer,
//     Sequential/layer.h:91-95); the reported loss is the per-sample L2 norm
//     of d_z (Sequential/Main.cpp:167-168)
//   * conv1 weight & bias grads normalized by 1/(24*24)
//     (Sequential/layer.h:381-389,398-414); pool/fc grads unnormalized
//   * pool bias grad averaged over its 216 outputs (Sequential/layer.h:302-317)
//   * update rule: p += dt * grad (gradient ascent on the residual ==
//     SGD on the squared error, Sequential/layer.h:97-101)
//
// Batched semantics: gradients are SUMS over the batch; the engine chooses
// the scale (1/B for mean reduction) at update time.  With B==1 and
// scale==1 a step is exactly one reference per-sample update.
//
// Parallelized race-free over the batch with at::parallel_for (OpenMP
// under the hood) — per-thread gradient scratch, reduced afterwards.
// This replaces the reference's racy OpenMP variant.

#include <ATen/Parallel.h>
#include <torch/extension.h>

#include <algorithm>
#include <cmath>
#include <cstring>
#include <vector>

#include "lenet_dims.h"

namespace pcnn {

static inline float sigmoidf(float v) { return 1.0f / (1.0f + std::exp(-v)); }

// pool_mode: 0 = trainable (reference), 1 = max.  loss_mode: 0 = residual
// (reference), 1 = softmax cross-entropy.

// Forward for one image: x[784] -> a1[3456], a2[216], y[10].
static void forward_one(const float* x, const float* p, float* a1, float* a2,
                        float* y, int pool_mode, int loss_mode) {
  const float* c1w = p + OFF_C1W;
  const float* c1b = p + OFF_C1B;
  const float* s1w = p + OFF_S1W;
  const float s1b = p[OFF_S1B];
  const float* fw = p + OFF_FW;
  const float* fb = p + OFF_FB;

  // conv1 + sigmoid
  for (int o = 0; o < C1_CH; ++o) {
    const float* w = c1w + o * C1_K * C1_K;
    for (int r = 0; r < C1_H; ++r) {
      for (int c = 0; c < C1_W; ++c) {
        float acc = c1b[o];
        for (int i = 0; i < C1_K; ++i)
          for (int j = 0; j < C1_K; ++j)
            acc += w[i * C1_K + j] * x[(r + i) * IN_W + (c + j)];
        a1[o * C1_PIX + r * C1_W + c] = sigmoidf(acc);
      }
    }
  }
  // pool + sigmoid: trainable shared 4x4 kernel (reference) or max
  for (int o = 0; o < C1_CH; ++o) {
    for (int pr = 0; pr < S1_H; ++pr) {
      for (int pc = 0; pc < S1_W; ++pc) {
        float acc;
        if (pool_mode == 1) {
          acc = -1e30f;
          for (int i = 0; i < S1_K; ++i)
            for (int j = 0; j < S1_K; ++j)
              acc = std::max(
                  acc,
                  a1[o * C1_PIX + (pr * S1_K + i) * C1_W + (pc * S1_K + j)]);
        } else {
          acc = s1b;
          for (int i = 0; i < S1_K; ++i)
            for (int j = 0; j < S1_K; ++j)
              acc += s1w[i * S1_K + j] *
                     a1[o * C1_PIX + (pr * S1_K + i) * C1_W +
                        (pc * S1_K + j)];
        }
        a2[o * S1_PIX + pr * S1_W + pc] = sigmoidf(acc);
      }
    }
  }
  // fc + sigmoid (residual) or softmax (cross-entropy)
  float z[FC_OUT];
  for (int k = 0; k < FC_OUT; ++k) {
    float acc = fb[k];
    const float* wk = fw + k * FC_IN;
    for (int m = 0; m < FC_IN; ++m) acc += wk[m] * a2[m];
    z[k] = acc;
  }
  if (loss_mode == 1) {
    float mx = z[0];
    for (int k = 1; k < FC_OUT; ++k) mx = std::max(mx, z[k]);
    float sum = 0.f;
    for (int k = 0; k < FC_OUT; ++k) {
      y[k] = std::exp(z[k] - mx);
      sum += y[k];
    }
    for (int k = 0; k < FC_OUT; ++k) y[k] /= sum;
  } else {
    for (int k = 0; k < FC_OUT; ++k) y[k] = sigmoidf(z[k]);
  }
}

// Backward for one image.  Fills dz[10], dz2[216], dz1[3456]; accumulates
// parameter gradients (sum) into g[N_PARAMS]; returns ||dz||_2.
static float backward_one(const float* x, const float* p, const float* a1,
                          const float* a2, const float* y, int64_t label,
                          float* dz, float* dz2, float* dz1, float* g,
                          int pool_mode, int loss_mode) {
  const float* s1w = p + OFF_S1W;
  const float* fw = p + OFF_FW;

  // residual / softmax-CE gradient: dz = onehot - y in both conventions
  float sq = 0.f;
  for (int k = 0; k < FC_OUT; ++k) {
    dz[k] = (k == label ? 1.0f : 0.0f) - y[k];
    sq += dz[k] * dz[k];
  }
  const float loss_v = loss_mode == 1
                           ? -std::log(std::max(y[label], 1e-30f))
                           : std::sqrt(sq);

  // fc wgrad / bgrad
  for (int k = 0; k < FC_OUT; ++k) {
    float* gw = g + OFF_FW + k * FC_IN;
    for (int m = 0; m < FC_IN; ++m) gw[m] += dz[k] * a2[m];
    g[OFF_FB + k] += dz[k];
  }

  // pool output grad -> preact grad
  float s1b_acc = 0.f;
  for (int m = 0; m < FC_IN; ++m) {
    float da = 0.f;
    for (int k = 0; k < FC_OUT; ++k) da += fw[k * FC_IN + m] * dz[k];
    float v = a2[m];
    dz2[m] = da * v * (1.0f - v);
    s1b_acc += dz2[m];
  }
  if (pool_mode == 0) {
    g[OFF_S1B] += s1b_acc / (float)S1_OUT;
    // pool wgrad (trainable only; max pool has no parameters)
    for (int i = 0; i < S1_K; ++i) {
      for (int j = 0; j < S1_K; ++j) {
        float acc = 0.f;
        for (int o = 0; o < C1_CH; ++o)
          for (int pr = 0; pr < S1_H; ++pr)
            for (int pc = 0; pc < S1_W; ++pc)
              acc +=
                  dz2[o * S1_PIX + pr * S1_W + pc] *
                  a1[o * C1_PIX + (pr * S1_K + i) * C1_W + (pc * S1_K + j)];
        g[OFF_S1W + i * S1_K + j] += acc;
      }
    }
  }

  // conv1 output grad -> preact grad.  Trainable pool: every position gets
  // dz2 * kernel weight; max pool: only the window argmax gets dz2.
  if (pool_mode == 1) {
    for (int o = 0; o < C1_CH; ++o) {
      for (int pr = 0; pr < S1_H; ++pr) {
        for (int pc = 0; pc < S1_W; ++pc) {
          int best = 0;
          float bv = -1e30f;
          for (int t = 0; t < S1_K * S1_K; ++t) {
            const int r = pr * S1_K + t / S1_K, c = pc * S1_K + t % S1_K;
            const float v = a1[o * C1_PIX + r * C1_W + c];
            if (v > bv) {
              bv = v;
              best = t;
            }
          }
          for (int t = 0; t < S1_K * S1_K; ++t) {
            const int r = pr * S1_K + t / S1_K, c = pc * S1_K + t % S1_K;
            const float v = a1[o * C1_PIX + r * C1_W + c];
            const float da =
                (t == best) ? dz2[o * S1_PIX + pr * S1_W + pc] : 0.f;
            dz1[o * C1_PIX + r * C1_W + c] = da * v * (1.0f - v);
          }
        }
      }
    }
  } else {
    for (int o = 0; o < C1_CH; ++o) {
      for (int r = 0; r < C1_H; ++r) {
        for (int c = 0; c < C1_W; ++c) {
          float da = dz2[o * S1_PIX + (r / S1_K) * S1_W + (c / S1_K)] *
                     s1w[(r % S1_K) * S1_K + (c % S1_K)];
          float v = a1[o * C1_PIX + r * C1_W + c];
          dz1[o * C1_PIX + r * C1_W + c] = da * v * (1.0f - v);
        }
      }
    }
  }

  // conv1 wgrad / bgrad, both normalized by 1/(24*24)
  constexpr float inv_pix = 1.0f / (float)C1_PIX;
  for (int o = 0; o < C1_CH; ++o) {
    float bacc = 0.f;
    for (int r = 0; r < C1_H; ++r)
      for (int c = 0; c < C1_W; ++c) bacc += dz1[o * C1_PIX + r * C1_W + c];
    g[OFF_C1B + o] += bacc * inv_pix;
    for (int i = 0; i < C1_K; ++i) {
      for (int j = 0; j < C1_K; ++j) {
        float acc = 0.f;
        for (int r = 0; r < C1_H; ++r)
          for (int c = 0; c < C1_W; ++c)
            acc += dz1[o * C1_PIX + r * C1_W + c] * x[(r + i) * IN_W + (c + j)];
        g[OFF_C1W + o * C1_K * C1_K + i * C1_K + j] += acc * inv_pix;
      }
    }
  }
  return loss_v;
}

static void check_cpu_f32(const at::Tensor& t, const char* name, int64_t numel) {
  TORCH_CHECK(t.device().is_cpu(), name, " must be a CPU tensor");
  TORCH_CHECK(t.scalar_type() == at::kFloat, name, " must be float32");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
  TORCH_CHECK(t.numel() == numel, name, " has wrong numel ", t.numel(),
              " expected ", numel);
}

void cpu_forward(at::Tensor x, at::Tensor params, at::Tensor a1, at::Tensor a2,
                 at::Tensor y, int64_t pool_mode, int64_t loss_mode) {
  int64_t B = x.size(0);
  check_cpu_f32(x, "x", B * IN_PIX);
  check_cpu_f32(params, "params", N_PARAMS);
  check_cpu_f32(a1, "a1", B * C1_OUT);
  check_cpu_f32(a2, "a2", B * S1_OUT);
  check_cpu_f32(y, "y", B * FC_OUT);
  const float* xp = x.data_ptr<float>();
  const float* pp = params.data_ptr<float>();
  float* a1p = a1.data_ptr<float>();
  float* a2p = a2.data_ptr<float>();
  float* yp = y.data_ptr<float>();
  at::parallel_for(0, B, 1, [&](int64_t b0, int64_t b1) {
    for (int64_t b = b0; b < b1; ++b)
      forward_one(xp + b * IN_PIX, pp, a1p + b * C1_OUT, a2p + b * S1_OUT,
                  yp + b * FC_OUT, (int)pool_mode, (int)loss_mode);
  });
}

// Returns the summed per-sample L2 loss norm; accumulates summed grads into
// `grads`.
double cpu_backward(at::Tensor x, at::Tensor params, at::Tensor a1,
                    at::Tensor a2, at::Tensor y, at::Tensor labels,
                    at::Tensor dz, at::Tensor dz2, at::Tensor dz1,
                    at::Tensor grads, int64_t pool_mode, int64_t loss_mode) {
  int64_t B = x.size(0);
  check_cpu_f32(x, "x", B * IN_PIX);
  check_cpu_f32(params, "params", N_PARAMS);
  check_cpu_f32(a1, "a1", B * C1_OUT);
  check_cpu_f32(a2, "a2", B * S1_OUT);
  check_cpu_f32(y, "y", B * FC_OUT);
  check_cpu_f32(dz, "dz", B * FC_OUT);
  check_cpu_f32(dz2, "dz2", B * S1_OUT);
  check_cpu_f32(dz1, "dz1", B * C1_OUT);
  check_cpu_f32(grads, "grads", N_PARAMS);
  TORCH_CHECK(labels.scalar_type() == at::kLong && labels.numel() == B,
              "labels must be int64 of shape [B]");
  const float* xp = x.data_ptr<float>();
  const float* pp = params.data_ptr<float>();
  const float* a1p = a1.data_ptr<float>();
  const float* a2p = a2.data_ptr<float>();
  const float* yp = y.data_ptr<float>();
  const int64_t* lp = labels.data_ptr<int64_t>();
  float* dzp = dz.data_ptr<float>();
  float* dz2p = dz2.data_ptr<float>();
  float* dz1p = dz1.data_ptr<float>();
  float* gp = grads.data_ptr<float>();

  int nt = at::get_num_threads();
  std::vector<std::vector<float>> scratch(nt);
  std::vector<double> losses(nt, 0.0);
  at::parallel_for(0, B, 1, [&](int64_t b0, int64_t b1) {
    int tid = at::get_thread_num();
    auto& g = scratch[tid];
    if (g.empty()) g.assign(N_PARAMS, 0.f);
    for (int64_t b = b0; b < b1; ++b) {
      losses[tid] += backward_one(xp + b * IN_PIX, pp, a1p + b * C1_OUT,
                                  a2p + b * S1_OUT, yp + b * FC_OUT, lp[b],
                                  dzp + b * FC_OUT, dz2p + b * S1_OUT,
                                  dz1p + b * C1_OUT, g.data(),
                                  (int)pool_mode, (int)loss_mode);
    }
  });
  double loss = 0.0;
  for (int t = 0; t < nt; ++t) {
    loss += losses[t];
    if (!scratch[t].empty())
      for (int i = 0; i < N_PARAMS; ++i) gp[i] += scratch[t][i];
  }
  return loss;
}

// p += dt * scale * g; g = 0.
void cpu_update(at::Tensor params, at::Tensor grads, double dt, double scale) {
  check_cpu_f32(params, "params", N_PARAMS);
  check_cpu_f32(grads, "grads", N_PARAMS);
  float* pp = params.data_ptr<float>();
  float* gp = grads.data_ptr<float>();
  const float s = (float)(dt * scale);
  for (int i = 0; i < N_PARAMS; ++i) {
    pp[i] += s * gp[i];
    gp[i] = 0.f;
  }
}

}  // namespace pcnn
