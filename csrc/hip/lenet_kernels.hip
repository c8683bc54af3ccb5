// Hand-written gfx950 (CDNA4 / MI355X) kernels for the LeNet-5-class
// training step.  Design notes:
//
// The whole network is ~1 MFLOP per image forward+backward, so on MI355X the
// bound is kernel-launch count and LATENCY (barrier phases, dependent memory
// chains), never FLOPs (SURVEY.md §7 "hard parts").  The reference CUDA
// variant spends 17 launches + 8 memsets per *sample* (SURVEY.md §2.3); here
// one training step of a whole batch is THREE kernels:
//
//   1. k_fwdbwd  — fused forward + backward-data.  One 256-thread workgroup
//      per image; activations live in LDS; 5 barrier phases
//      ({stage x+params} {conv1} {pool} {fc+loss-residual} {fc-bwd}
//      {pool-bwd}); the fc dot products are 16-lane shuffle reductions so no
//      phase has a >32-step dependency chain.  Fuses what the reference ran
//      as 12 separate kernels per sample.
//   2. k_wgrad   — all weight/bias gradients, batch-reduced.  Measured
//      lesson (profiles/, round 1): a chunk-serial design with LDS staging
//      and per-image barriers ran 62 us — latency-chained.  v2 is
//      barrier-free: grid-stride over flattened (image, position) work
//      items, per-thread register accumulators, wave-level __shfl_down
//      reduction, then a handful of hardware fp32 atomics
//      (unsafeAtomicAdd -> global_atomic_add_f32) into the flat gradient
//      bucket.  The fc weight grads use exclusive per-thread ownership of
//      (k,m) pairs — no atomics, plain accumulate-stores.
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

// Vectorized loads of 4/8 consecutive activations as fp32.  Callers
// guarantee 8-byte (bf16) / 16-byte (fp32) alignment of p.
__device__ __forceinline__ void ld4f(const bf16* p, float* out) {
  const ushort4 v = *reinterpret_cast<const ushort4*>(p);
  const bf16* e = reinterpret_cast<const bf16*>(&v);
  out[0] = (float)e[0];
  out[1] = (float)e[1];
  out[2] = (float)e[2];
  out[3] = (float)e[3];
}
__device__ __forceinline__ void ld4f(const float* p, float* out) {
  const float4 v = *reinterpret_cast<const float4*>(p);
  out[0] = v.x;
  out[1] = v.y;
  out[2] = v.z;
  out[3] = v.w;
}
template <typename T>
__device__ __forceinline__ void ld8f(const T* p, float* out) {
  ld4f(p, out);
  ld4f(p + 4, out + 4);
}

// Execution modes for the fused forward kernel.
enum Mode { MODE_TRAIN = 0, MODE_EVAL = 1, MODE_INFER = 2 };

// LDS image of the per-image state.  One single __shared__ object.
// Only the conv/pool parameters (173 floats) are staged; the fc weight
// matrix stays in global memory (L2-resident, read coalesced once per
// phase) — staging it cost a full extra LDS round and 9.4 KB of occupancy.
struct FwdLds {
  float ps[OFF_FW];     // staged conv1+pool parameters
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

  // ---- phase 0: stage conv/pool parameters and the input image ----
  if (tid < OFF_FW) L.ps[tid] = params[tid];
  const act_t* xb = x + (size_t)b * IN_PIX;
  for (int i = tid; i < IN_PIX; i += 256) L.xs[i] = ldf(xb + i);
  __syncthreads();

  // ---- phase 1: conv1 (5x5 valid, 6 ch) + sigmoid ----
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

  // ---- phase 2: trainable pool (4x4 stride 4, shared kernel) + sigmoid ----
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

  // ---- phase 3: fc (216 -> 10) + sigmoid [+ loss residual] ----
  // 16 lanes per output class: lane l of group k sums m = l, l+16, ...
  // then a 4-step shuffle tree; the group leader applies bias+sigmoid and
  // (TRAIN) computes the residual immediately — no extra barrier phase.
  if (tid < FC_OUT * 16) {
    const int k = tid >> 4;
    const int l = tid & 15;
    const float* wk = params + OFF_FW + k * FC_IN;
    float p = 0.f;
#pragma unroll
    for (int u = 0; u < (FC_IN + 15) / 16; ++u) {
      const int m = l + u * 16;
      if (m < FC_IN) p += wk[m] * L.a2s[m];
    }
#pragma unroll
    for (int off = 8; off > 0; off >>= 1) p += __shfl_down(p, off, 16);
    if (l == 0) {
      const float v = sigmoidf_dev(p + params[OFF_FB + k]);
      L.ys[k] = v;
      if (yg != nullptr) yg[(size_t)b * FC_OUT + k] = v;
      if (MODE == MODE_TRAIN) {
        const float d = (k == labels[b] ? 1.0f : 0.0f) - v;
        L.dzs[k] = d;
        L.sq[k] = d * d;
        dzg[(size_t)b * FC_OUT + k] = d;
      }
    }
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

  // ---- phase 4: fc backward-data -> pool preact gradient ----
  if (tid < S1_OUT) {
    float da = 0.f;
#pragma unroll
    for (int k = 0; k < FC_OUT; ++k)
      da += params[OFF_FW + k * FC_IN + tid] * L.dzs[k];
    const float v = L.a2s[tid];
    const float d = da * v * (1.0f - v);
    L.dz2s[tid] = d;
    dz2g[(size_t)b * S1_OUT + tid] = d;
  } else if (tid == S1_OUT && loss_accum != nullptr) {
    float s = 0.f;
#pragma unroll
    for (int k = 0; k < FC_OUT; ++k) s += L.sq[k];
    unsafeAtomicAdd(loss_accum, sqrtf(s));
  }
  __syncthreads();

  // ---- phase 5: pool backward-data -> conv1 preact gradient ----
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
// Weight gradients, batch-reduced (v2 — barrier-free grid-stride).
//
// Grid role layout (GC = blocks per conv channel, GS = pool blocks):
//   blocks [0, 6*GC)            : conv1 wgrad+bgrad, blk -> (channel, slice)
//   blocks [6*GC, 6*GC+GS)      : pool wgrad+bgrad
//   blocks [6*GC+GS, +9)        : fc wgrad+bgrad (2160 weights + 10 biases,
//                                 one owner thread each — no atomics)
// ---------------------------------------------------------------------------

constexpr int WG_GC_DEFAULT = 8;   // conv1 slices per channel (runtime knob)
constexpr int FC_BLOCKS = (FC_WSZ + 255) / 256;  // 9

// Cross-lane sum over the full 64-lane wave.
__device__ __forceinline__ float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
  return v;
}

template <typename act_t>
__global__ __launch_bounds__(256) void k_wgrad(
    const act_t* __restrict__ x, const act_t* __restrict__ a1g,
    const act_t* __restrict__ a2g, const float* __restrict__ dzg,
    const float* __restrict__ dz2g, const float* __restrict__ dz1g,
    float* __restrict__ grads, int B, int GC, int GS, int FS, int roles) {
  const int tid = threadIdx.x;
  const int blk = blockIdx.x;
  const int lane = tid & 63;

  if (blk < C1_CH * GC) {
    if (!(roles & 1)) return;
    // ---- conv1: dW[o,i,j] = sum_{b,r,c} dz1[b,o,r,c] * x[b,r+i,c+j] / 576
    // Work item = 4 consecutive output columns of one row: 5 vectorized
    // 8-wide x row loads + one float4 dz1 load replace 104 scalar loads
    // (the v2 scalar form was address-issue-bound: profiles/ r1).
    const int o = blk / GC;
    const int slice = blk - o * GC;
    constexpr int GROUPS_PER_ROW = C1_W / 4;  // 6
    float acc[C1_K * C1_K];
#pragma unroll
    for (int w = 0; w < C1_K * C1_K; ++w) acc[w] = 0.f;
    float bacc = 0.f;
    const int N = B * C1_H * GROUPS_PER_ROW;
    for (int it = slice * 256 + tid; it < N; it += GC * 256) {
      const int b = it / (C1_H * GROUPS_PER_ROW);
      const int rg = it - b * (C1_H * GROUPS_PER_ROW);
      const int r = rg / GROUPS_PER_ROW;
      const int c0 = (rg - r * GROUPS_PER_ROW) * 4;
      float d4[4];
      ld4f(dz1g + (size_t)b * C1_OUT + o * C1_PIX + r * C1_W + c0, d4);
      bacc += d4[0] + d4[1] + d4[2] + d4[3];
      const act_t* xb = x + (size_t)b * IN_PIX + r * IN_W + c0;
      float xr[8];
#pragma unroll
      for (int i = 0; i < C1_K; ++i) {
        ld8f(xb + i * IN_W, xr);
#pragma unroll
        for (int pp = 0; pp < 4; ++pp)
#pragma unroll
          for (int j = 0; j < C1_K; ++j)
            acc[i * C1_K + j] += d4[pp] * xr[pp + j];
      }
    }
    constexpr float inv_pix = 1.0f / (float)C1_PIX;
#pragma unroll
    for (int w = 0; w < C1_K * C1_K; ++w) {
      const float s = wave_sum(acc[w]);
      if (lane == 0)
        unsafeAtomicAdd(&grads[OFF_C1W + o * C1_K * C1_K + w], s * inv_pix);
    }
    const float bs = wave_sum(bacc);
    if (lane == 0) unsafeAtomicAdd(&grads[OFF_C1B + o], bs * inv_pix);
  } else if (blk < C1_CH * GC + GS) {
    if (!(roles & 2)) return;
    // ---- pool: dW[i,j] = sum_{b,o,p,q} dz2[b,o,p,q] * a1[b,o,4p+i,4q+j]
    const int slice = blk - C1_CH * GC;
    float acc[S1_WSZ];
#pragma unroll
    for (int w = 0; w < S1_WSZ; ++w) acc[w] = 0.f;
    float bacc = 0.f;
    const int N = B * S1_OUT;
    for (int it = slice * 256 + tid; it < N; it += GS * 256) {
      const int b = it / S1_OUT;
      const int opq = it - b * S1_OUT;
      const int o = opq / S1_PIX;
      const int pq = opq - o * S1_PIX;
      const int pr = pq / S1_W;
      const int pc = pq - pr * S1_W;
      const float d = dz2g[it];
      bacc += d;
      const act_t* base = a1g + (size_t)b * C1_OUT + o * C1_PIX +
                          pr * S1_K * C1_W + pc * S1_K;
      float ar[4];
#pragma unroll
      for (int i = 0; i < S1_K; ++i) {
        ld4f(base + i * C1_W, ar);
#pragma unroll
        for (int j = 0; j < S1_K; ++j) acc[i * S1_K + j] += d * ar[j];
      }
    }
#pragma unroll
    for (int w = 0; w < S1_WSZ; ++w) {
      const float s = wave_sum(acc[w]);
      if (lane == 0) unsafeAtomicAdd(&grads[OFF_S1W + w], s);
    }
    const float bs = wave_sum(bacc);
    if (lane == 0)
      unsafeAtomicAdd(&grads[OFF_S1B], bs / (float)S1_OUT);
  } else {
    if (!(roles & 4)) return;
    // ---- fc: dW[k,m] = sum_b dz[b,k] * a2[b,m];  db[k] = sum_b dz[b,k]
    // FS batch slices; one owner thread per (weight, slice).  With FS==1
    // the owner accumulates with a plain store (no atomics); FS>1 (large
    // batch) uses hardware atomics across slices.
    const int fblk = blk - C1_CH * GC - GS;
    const int slice = fblk / FC_BLOCKS;
    const int q = (fblk - slice * FC_BLOCKS) * 256 + tid;
    const int b_lo = (int)(((long)B * slice) / FS);
    const int b_hi = (int)(((long)B * (slice + 1)) / FS);
    if (q < FC_WSZ) {
      const int k = q / FC_IN;
      const int m = q - k * FC_IN;
      float acc = 0.f;
#pragma unroll 8
      for (int b = b_lo; b < b_hi; ++b)
        acc += dzg[(size_t)b * FC_OUT + k] * ldf(a2g + (size_t)b * FC_IN + m);
      if (FS == 1)
        grads[OFF_FW + q] += acc;
      else
        unsafeAtomicAdd(&grads[OFF_FW + q], acc);
    } else if (q < FC_WSZ + FC_OUT) {
      const int k = q - FC_WSZ;
      float acc = 0.f;
#pragma unroll 8
      for (int b = b_lo; b < b_hi; ++b) acc += dzg[(size_t)b * FC_OUT + k];
      if (FS == 1)
        grads[OFF_FB + k] += acc;
      else
        unsafeAtomicAdd(&grads[OFF_FB + k], acc);
    }
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

// chunk_imgs is the conv1 slices-per-channel knob (GC); <=0 -> default.
int pcnn_launch_wgrad_ex(const void* x, const void* a1, const void* a2,
                         const float* dz, const float* dz2, const float* dz1,
                         float* grads, int B, int act_is_bf16, int chunk_imgs,
                         int roles, void* stream) {
  // Batch-adaptive defaults: ~36 (image,position) items per conv thread,
  // pool slices at GC/4, fc batch slices of ~64 images.
  int GC = chunk_imgs > 0 ? chunk_imgs : (int)(2.0f * __builtin_cbrtf((float)B) + 0.5f);
  if (GC < 2) GC = 2;
  if (GC > 256) GC = 256;
  int GS = GC / 4 > 2 ? GC / 4 : 2;
  int FS = B / 64;
  if (FS < 1) FS = 1;
  if (FS > 32) FS = 32;
  dim3 grid(C1_CH * GC + GS + FS * FC_BLOCKS), block(256);
  hipStream_t s = (hipStream_t)stream;
  if (act_is_bf16) {
    hipLaunchKernelGGL((k_wgrad<bf16>), grid, block, 0, s, (const bf16*)x,
                       (const bf16*)a1, (const bf16*)a2, dz, dz2, dz1, grads,
                       B, GC, GS, FS, roles);
  } else {
    hipLaunchKernelGGL((k_wgrad<float>), grid, block, 0, s, (const float*)x,
                       (const float*)a1, (const float*)a2, dz, dz2, dz1, grads,
                       B, GC, GS, FS, roles);
  }
  return (int)hipGetLastError();
}

int pcnn_launch_wgrad(const void* x, const void* a1, const void* a2,
                      const float* dz, const float* dz2, const float* dz1,
                      float* grads, int B, int act_is_bf16, int chunk_imgs,
                      void* stream) {
  return pcnn_launch_wgrad_ex(x, a1, a2, dz, dz2, dz1, grads, B, act_is_bf16,
                              chunk_imgs, 7, stream);
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
