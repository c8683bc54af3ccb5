// Hand-written gfx950 (CDNA4 / MI355X) kernels for the LeNet-5-class
// training step.  Design notes:
//
// The whole network is ~1 MFLOP per image forward+backward, so on MI355X the
// bound is kernel-launch count and memory latency, never FLOPs (SURVEY.md §7
// "hard parts").  The reference CUDA variant spends 17 launches + 8 memsets
// per *sample* (SURVEY.md §2.3); here one training step of a whole batch is
// THREE kernels:
//
//   1. k_fwdbwd  — fused forward + backward-data.  One 256-thread workgroup
//      per image; every activation lives in LDS (≈28 KB incl. a parameter
//      stage), global traffic is one read of x and one write of the tensors
//      the weight-grad kernel needs.  Fuses what the reference ran as 12
//      separate kernels (fp_c1, sigmoid, fp_s1, sigmoid, fp_f, sigmoid,
//      makeError, nrm2, bp_output_s1, bp_preact_s1, bp_output_c1,
//      bp_preact_c1).
//   2. k_wgrad   — all weight/bias gradients, batch-reduced.  Grid is
//      (role × batch-chunk); per-thread register accumulation over the
//      chunk, one fp32 atomicAdd per weight per block into the flat
//      gradient bucket (which is also the RCCL all-reduce payload).
//   3. k_update  — SGD apply (p += dt*scale*g) fused with gradient zeroing.
//
// Numerics: activations are stored bf16 (or fp32, template) in global
// memory; ALL arithmetic is fp32 in registers/LDS; parameters, gradients and
// backward-data tensors are fp32.  Loss-metric semantics match the reference
// (sum over samples of ||onehot - y||_2).
//
// Wave width is 64 (CDNA4); block size 256 = 4 waves.

#include <hip/hip_bf16.h>
#include <hip/hip_runtime.h>

#include "../lenet_dims.h"

namespace pcnn {

using bf16 = __hip_bfloat16;

__device__ __forceinline__ float sigmoidf_dev(float v) {
  return 1.0f / (1.0f + __expf(-v));
}

template <typename T>
__device__ __forceinline__ float ldf(const T* p) {
  return (float)*p;
}
template <typename T>
__device__ __forceinline__ void stf(T* p, float v) {
  *p = (T)v;
}

// Execution modes for the fused forward kernel.
enum Mode { MODE_TRAIN = 0, MODE_EVAL = 1, MODE_INFER = 2 };

// LDS image of the per-image state.  One single __shared__ object.
struct FwdLds {
  float ps[N_PARAMS];   // staged parameters
  float xs[IN_PIX];     // input image (fp32)
  float a1s[C1_OUT];    // conv1 activation
  float a2s[S1_OUT];    // pool activation
  float ys[FC_OUT];     // logits (post-sigmoid)
  float dzs[FC_OUT];    // residual gradient
  float sq[FC_OUT];     // per-class squared error
  float dz2s[S1_OUT];   // pool preact gradient
};

template <typename act_t, int MODE>
__global__ __launch_bounds__(256) void k_fwdbwd(
    const act_t* __restrict__ x, const float* __restrict__ params,
    act_t* __restrict__ a1g, act_t* __restrict__ a2g, float* __restrict__ yg,
    float* __restrict__ dzg, float* __restrict__ dz2g,
    float* __restrict__ dz1g, const int* __restrict__ labels,
    float* __restrict__ loss_accum, int* __restrict__ correct_accum, int B) {
  __shared__ FwdLds L;
  const int b = blockIdx.x;
  if (b >= B) return;
  const int tid = threadIdx.x;

  // Stage parameters and the input image into LDS.
  for (int i = tid; i < N_PARAMS; i += 256) L.ps[i] = params[i];
  const act_t* xb = x + (size_t)b * IN_PIX;
  for (int i = tid; i < IN_PIX; i += 256) L.xs[i] = ldf(xb + i);
  __syncthreads();

  // ---- conv1 (5x5 valid, 6 ch) + sigmoid ----
  for (int t = tid; t < C1_OUT; t += 256) {
    const int o = t / C1_PIX;
    const int rc = t - o * C1_PIX;
    const int r = rc / C1_W;
    const int c = rc - r * C1_W;
    const float* w = &L.ps[OFF_C1W + o * C1_K * C1_K];
    float acc = L.ps[OFF_C1B + o];
#pragma unroll
    for (int i = 0; i < C1_K; ++i)
#pragma unroll
      for (int j = 0; j < C1_K; ++j)
        acc += w[i * C1_K + j] * L.xs[(r + i) * IN_W + (c + j)];
    const float v = sigmoidf_dev(acc);
    L.a1s[t] = v;
    if (MODE == MODE_TRAIN) stf(a1g + (size_t)b * C1_OUT + t, v);
  }
  __syncthreads();

  // ---- trainable pool (4x4 stride 4, shared kernel) + sigmoid ----
  if (tid < S1_OUT) {
    const int o = tid / S1_PIX;
    const int pq = tid - o * S1_PIX;
    const int pr = pq / S1_W;
    const int pc = pq - pr * S1_W;
    const float* base = &L.a1s[o * C1_PIX + pr * S1_K * C1_W + pc * S1_K];
    float acc = L.ps[OFF_S1B];
#pragma unroll
    for (int i = 0; i < S1_K; ++i)
#pragma unroll
      for (int j = 0; j < S1_K; ++j)
        acc += L.ps[OFF_S1W + i * S1_K + j] * base[i * C1_W + j];
    const float v = sigmoidf_dev(acc);
    L.a2s[tid] = v;
    if (MODE == MODE_TRAIN) stf(a2g + (size_t)b * S1_OUT + tid, v);
  }
  __syncthreads();

  // ---- fc (216 -> 10) + sigmoid ----
  if (tid < FC_OUT) {
    float acc = L.ps[OFF_FB + tid];
    const float* wk = &L.ps[OFF_FW + tid * FC_IN];
#pragma unroll 8
    for (int m = 0; m < FC_IN; ++m) acc += wk[m] * L.a2s[m];
    const float v = sigmoidf_dev(acc);
    L.ys[tid] = v;
    if (yg != nullptr) yg[(size_t)b * FC_OUT + tid] = v;
  }
  __syncthreads();

  if (MODE == MODE_EVAL) {
    // argmax + correct-count (replaces the reference's per-image D2H copy +
    // host argmax, CUDA/main.cu:220)
    if (tid == 0) {
      int best = 0;
      for (int k = 1; k < FC_OUT; ++k)
        if (L.ys[k] > L.ys[best]) best = k;
      if (best == labels[b]) atomicAdd(correct_accum, 1);
    }
    return;
  }
  if (MODE == MODE_INFER) return;

  // ---- residual loss gradient + loss metric ----
  if (tid < FC_OUT) {
    const float d = (tid == labels[b] ? 1.0f : 0.0f) - L.ys[tid];
    L.dzs[tid] = d;
    L.sq[tid] = d * d;
    dzg[(size_t)b * FC_OUT + tid] = d;
  }
  __syncthreads();
  if (tid == 0 && loss_accum != nullptr) {
    float s = 0.f;
#pragma unroll
    for (int k = 0; k < FC_OUT; ++k) s += L.sq[k];
    unsafeAtomicAdd(loss_accum, sqrtf(s));
  }

  // ---- fc backward-data -> pool preact gradient ----
  if (tid < S1_OUT) {
    float da = 0.f;
#pragma unroll
    for (int k = 0; k < FC_OUT; ++k)
      da += L.ps[OFF_FW + k * FC_IN + tid] * L.dzs[k];
    const float v = L.a2s[tid];
    const float d = da * v * (1.0f - v);
    L.dz2s[tid] = d;
    dz2g[(size_t)b * S1_OUT + tid] = d;
  }
  __syncthreads();

  // ---- pool backward-data -> conv1 preact gradient ----
  // stride == kernel: each conv1 output feeds exactly one pool cell (gather).
  for (int t = tid; t < C1_OUT; t += 256) {
    const int o = t / C1_PIX;
    const int rc = t - o * C1_PIX;
    const int r = rc / C1_W;
    const int c = rc - r * C1_W;
    const float da = L.dz2s[o * S1_PIX + (r / S1_K) * S1_W + (c / S1_K)] *
                     L.ps[OFF_S1W + (r % S1_K) * S1_K + (c % S1_K)];
    const float v = L.a1s[t];
    dz1g[(size_t)b * C1_OUT + t] = da * v * (1.0f - v);
  }
}

// ---------------------------------------------------------------------------
// Weight gradients, batch-reduced.
//
// Grid role layout with NC = ceil(B / CHUNK) batch-chunks:
//   blocks [0, 6*NC)      : conv1 wgrad+bgrad, one block per (channel, chunk)
//   blocks [6*NC, 7*NC)   : pool wgrad+bgrad, one block per chunk
//   blocks [7*NC, 8*NC)   : fc wgrad+bgrad, one block per chunk
// Each block accumulates over its chunk in registers, then atomicAdds into
// the flat fp32 gradient bucket (conflicts only across chunks).
// ---------------------------------------------------------------------------

constexpr int WG_CHUNK_DEFAULT = 8;  // images per chunk (runtime-tunable)

template <typename act_t>
__global__ __launch_bounds__(256) void k_wgrad(
    const act_t* __restrict__ x, const act_t* __restrict__ a1g,
    const act_t* __restrict__ a2g, const float* __restrict__ dzg,
    const float* __restrict__ dz2g, const float* __restrict__ dz1g,
    float* __restrict__ grads, int B, int NC, int CIMG) {
  const int tid = threadIdx.x;
  const int blk = blockIdx.x;

  if (blk < C1_CH * NC) {
    // ---- conv1: dW[o,i,j] = sum_{b,r,c} dz1[b,o,r,c] * x[b,r+i,c+j] / 576
    const int o = blk / NC;
    const int chunk = blk - o * NC;
    const int b_lo = chunk * CIMG;
    const int b_hi = min(B, b_lo + CIMG);
    __shared__ float S[IN_PIX + C1_K * C1_K + 1];
    float* xs = S;
    float* wacc = S + IN_PIX;     // [25] block-level accumulators
    float* bacc = wacc + C1_K * C1_K;
    if (tid < C1_K * C1_K + 1) wacc[tid] = 0.f;
    float acc[C1_K * C1_K];
#pragma unroll
    for (int w = 0; w < C1_K * C1_K; ++w) acc[w] = 0.f;
    float bsum = 0.f;
    for (int b = b_lo; b < b_hi; ++b) {
      __syncthreads();  // protect xs reload
      const act_t* xb = x + (size_t)b * IN_PIX;
      for (int i = tid; i < IN_PIX; i += 256) xs[i] = ldf(xb + i);
      __syncthreads();
      const float* dz1b = dz1g + (size_t)b * C1_OUT + o * C1_PIX;
      for (int t = tid; t < C1_PIX; t += 256) {
        const int r = t / C1_W;
        const int c = t - r * C1_W;
        const float d = dz1b[t];
        bsum += d;
#pragma unroll
        for (int i = 0; i < C1_K; ++i)
#pragma unroll
          for (int j = 0; j < C1_K; ++j)
            acc[i * C1_K + j] += d * xs[(r + i) * IN_W + (c + j)];
      }
    }
    __syncthreads();
#pragma unroll
    for (int w = 0; w < C1_K * C1_K; ++w) atomicAdd(&wacc[w], acc[w]);
    atomicAdd(bacc, bsum);
    __syncthreads();
    constexpr float inv_pix = 1.0f / (float)C1_PIX;
    if (tid < C1_K * C1_K)
      unsafeAtomicAdd(&grads[OFF_C1W + o * C1_K * C1_K + tid], wacc[tid] * inv_pix);
    if (tid == C1_K * C1_K) unsafeAtomicAdd(&grads[OFF_C1B + o], bacc[0] * inv_pix);
  } else if (blk < (C1_CH + 1) * NC) {
    // ---- pool: dW[i,j] = sum_{b,o,p,q} dz2[b,o,p,q] * a1[b,o,4p+i,4q+j]
    const int chunk = blk - C1_CH * NC;
    const int b_lo = chunk * CIMG;
    const int b_hi = min(B, b_lo + CIMG);
    __shared__ float S[C1_OUT + S1_WSZ + 1];
    float* a1s = S;
    float* wacc = S + C1_OUT;
    float* bacc = wacc + S1_WSZ;
    if (tid < S1_WSZ + 1) wacc[tid] = 0.f;
    float acc[S1_WSZ];
#pragma unroll
    for (int w = 0; w < S1_WSZ; ++w) acc[w] = 0.f;
    float bsum = 0.f;
    for (int b = b_lo; b < b_hi; ++b) {
      __syncthreads();
      const act_t* a1b = a1g + (size_t)b * C1_OUT;
      for (int i = tid; i < C1_OUT; i += 256) a1s[i] = ldf(a1b + i);
      __syncthreads();
      if (tid < S1_OUT) {
        const int o = tid / S1_PIX;
        const int pq = tid - o * S1_PIX;
        const int pr = pq / S1_W;
        const int pc = pq - pr * S1_W;
        const float d = dz2g[(size_t)b * S1_OUT + tid];
        bsum += d;
        const float* base = &a1s[o * C1_PIX + pr * S1_K * C1_W + pc * S1_K];
#pragma unroll
        for (int i = 0; i < S1_K; ++i)
#pragma unroll
          for (int j = 0; j < S1_K; ++j)
            acc[i * S1_K + j] += d * base[i * C1_W + j];
      }
    }
    __syncthreads();
#pragma unroll
    for (int w = 0; w < S1_WSZ; ++w) atomicAdd(&wacc[w], acc[w]);
    atomicAdd(bacc, bsum);
    __syncthreads();
    if (tid < S1_WSZ) unsafeAtomicAdd(&grads[OFF_S1W + tid], wacc[tid]);
    if (tid == S1_WSZ)
      unsafeAtomicAdd(&grads[OFF_S1B], bacc[0] / (float)S1_OUT);
  } else {
    // ---- fc: dW[k,m] = sum_b dz[b,k] * a2[b,m];  db[k] = sum_b dz[b,k]
    const int chunk = blk - (C1_CH + 1) * NC;
    const int b_lo = chunk * CIMG;
    const int b_hi = min(B, b_lo + CIMG);
    __shared__ float S[S1_OUT + FC_OUT];
    float* a2s = S;
    float* dzs = S + S1_OUT;
    constexpr int NACC = (FC_WSZ + 255) / 256;  // 9 (k,m) pairs per thread
    float acc[NACC];
#pragma unroll
    for (int u = 0; u < NACC; ++u) acc[u] = 0.f;
    float bsum = 0.f;  // threads 0..9 hold fc bias grad
    for (int b = b_lo; b < b_hi; ++b) {
      __syncthreads();
      const act_t* a2b = a2g + (size_t)b * S1_OUT;
      if (tid < S1_OUT) a2s[tid] = ldf(a2b + tid);
      if (tid >= S1_OUT && tid < S1_OUT + FC_OUT)
        dzs[tid - S1_OUT] = dzg[(size_t)b * FC_OUT + (tid - S1_OUT)];
      __syncthreads();
#pragma unroll
      for (int u = 0; u < NACC; ++u) {
        const int q = tid + u * 256;
        if (q < FC_WSZ) {
          const int k = q / FC_IN;
          const int m = q - k * FC_IN;
          acc[u] += dzs[k] * a2s[m];
        }
      }
      if (tid < FC_OUT) bsum += dzs[tid];
    }
#pragma unroll
    for (int u = 0; u < NACC; ++u) {
      const int q = tid + u * 256;
      if (q < FC_WSZ) unsafeAtomicAdd(&grads[OFF_FW + q], acc[u]);
    }
    if (tid < FC_OUT) unsafeAtomicAdd(&grads[OFF_FB + tid], bsum);
  }
}

// ---------------------------------------------------------------------------
// SGD apply + gradient zeroing:  p += dt*scale*g;  g = 0.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void k_update(float* __restrict__ params,
                                                float* __restrict__ grads,
                                                float step) {
  const int i = blockIdx.x * 256 + threadIdx.x;
  if (i < N_PARAMS) {
    params[i] += step * grads[i];
    grads[i] = 0.f;
  }
}

}  // namespace pcnn

// ---------------------------------------------------------------------------
// extern "C" launchers (called from the pybind layer; no torch headers here).
// All return hipError_t as int.
// ---------------------------------------------------------------------------

using namespace pcnn;

namespace {
template <int MODE>
int launch_fwdbwd_mode(const void* x, const float* params, void* a1, void* a2,
                       float* y, float* dz, float* dz2, float* dz1,
                       const int* labels, float* loss_accum, int* correct,
                       int B, int act_is_bf16, hipStream_t stream) {
  dim3 grid(B), block(256);
  if (act_is_bf16) {
    hipLaunchKernelGGL((k_fwdbwd<bf16, MODE>), grid, block, 0, stream,
                       (const bf16*)x, params, (bf16*)a1, (bf16*)a2, y, dz,
                       dz2, dz1, labels, loss_accum, correct, B);
  } else {
    hipLaunchKernelGGL((k_fwdbwd<float, MODE>), grid, block, 0, stream,
                       (const float*)x, params, (float*)a1, (float*)a2, y, dz,
                       dz2, dz1, labels, loss_accum, correct, B);
  }
  return (int)hipGetLastError();
}
}  // namespace

extern "C" {

int pcnn_launch_fwdbwd(const void* x, const float* params, void* a1, void* a2,
                       float* y, float* dz, float* dz2, float* dz1,
                       const int* labels, float* loss_accum, int* correct,
                       int B, int act_is_bf16, int mode, void* stream) {
  hipStream_t s = (hipStream_t)stream;
  switch (mode) {
    case MODE_TRAIN:
      return launch_fwdbwd_mode<MODE_TRAIN>(x, params, a1, a2, y, dz, dz2, dz1,
                                            labels, loss_accum, correct, B,
                                            act_is_bf16, s);
    case MODE_EVAL:
      return launch_fwdbwd_mode<MODE_EVAL>(x, params, a1, a2, y, dz, dz2, dz1,
                                           labels, loss_accum, correct, B,
                                           act_is_bf16, s);
    default:
      return launch_fwdbwd_mode<MODE_INFER>(x, params, a1, a2, y, dz, dz2, dz1,
                                            labels, loss_accum, correct, B,
                                            act_is_bf16, s);
  }
}

int pcnn_launch_wgrad(const void* x, const void* a1, const void* a2,
                      const float* dz, const float* dz2, const float* dz1,
                      float* grads, int B, int act_is_bf16, int chunk_imgs,
                      void* stream) {
  const int CIMG = chunk_imgs > 0 ? chunk_imgs : WG_CHUNK_DEFAULT;
  const int NC = (B + CIMG - 1) / CIMG;
  dim3 grid((C1_CH + 2) * NC), block(256);
  hipStream_t s = (hipStream_t)stream;
  if (act_is_bf16) {
    hipLaunchKernelGGL((k_wgrad<bf16>), grid, block, 0, s, (const bf16*)x,
                       (const bf16*)a1, (const bf16*)a2, dz, dz2, dz1, grads,
                       B, NC, CIMG);
  } else {
    hipLaunchKernelGGL((k_wgrad<float>), grid, block, 0, s, (const float*)x,
                       (const float*)a1, (const float*)a2, dz, dz2, dz1, grads,
                       B, NC, CIMG);
  }
  return (int)hipGetLastError();
}

int pcnn_launch_update(float* params, float* grads, float step, void* stream) {
  dim3 grid((N_PARAMS + 255) / 256), block(256);
  hipLaunchKernelGGL(k_update, grid, block, 0, (hipStream_t)stream, params,
                     grads, step);
  return (int)hipGetLastError();
}

const char* pcnn_hip_error_string(int err) {
  return hipGetErrorString((hipError_t)err);
}
}
