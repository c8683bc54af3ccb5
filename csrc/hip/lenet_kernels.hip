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
//      per image; 3 barriers total (thread t owns pool cell t end-to-end:
//      its 16 conv outputs stay in registers from forward into the pool
//      backward); the fc dot products are 16-lane shuffle reductions so no
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
// Numerics: activations AND the large conv preact gradient (dz1) are
// stored bf16/fp16/fp32 (template); ALL arithmetic is fp32 in
// registers/LDS; parameters, gradients, dz and dz2 are fp32.  Loss-metric
// semantics match the reference (sum over samples of ||onehot - y||_2,
// SURVEY.md §0.1 item 3).
//
// Wave width is 64 (CDNA4); block size 256 = 4 waves.

#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <hip/hip_runtime.h>

#include "../lenet_dims.h"

namespace pcnn {

using bf16 = __hip_bfloat16;
using fp16 = __half;

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
__device__ __forceinline__ void ld4f(const fp16* p, float* out) {
  const ushort4 v = *reinterpret_cast<const ushort4*>(p);
  const fp16* e = reinterpret_cast<const fp16*>(&v);
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

// Vectorized store of 4 consecutive activations (8-byte-aligned bf16 /
// 16-byte-aligned fp32 destination).
__device__ __forceinline__ void st4f(bf16* p, const float* v) {
  bf16 t[4] = {(bf16)v[0], (bf16)v[1], (bf16)v[2], (bf16)v[3]};
  *reinterpret_cast<uint2*>(p) = *reinterpret_cast<const uint2*>(t);
}
__device__ __forceinline__ void st4f(fp16* p, const float* v) {
  fp16 t[4] = {(fp16)v[0], (fp16)v[1], (fp16)v[2], (fp16)v[3]};
  *reinterpret_cast<uint2*>(p) = *reinterpret_cast<const uint2*>(t);
}
__device__ __forceinline__ void st4f(float* p, const float* v) {
  *reinterpret_cast<float4*>(p) = make_float4(v[0], v[1], v[2], v[3]);
}

// Execution modes for the fused forward kernel.
enum Mode { MODE_TRAIN = 0, MODE_EVAL = 1, MODE_INFER = 2 };

// LDS image of the per-image state.  One single __shared__ object.
// Only the conv/pool parameters (173 floats) are staged; the fc weight
// matrix stays in global memory (L2-resident, read coalesced once per
// phase).
struct FwdLds {
  float a2s[S1_OUT];    // pool activation
  float ys[FC_OUT];     // logits (post-sigmoid)
  float dzs[FC_OUT];    // residual gradient
  float sq[FC_OUT];     // per-class squared error
  // fused weight-grad accumulators (wgrad_fuse mode): conv1 per-channel
  // 5x5 + bias, pool 4x4 + bias
  float gw[C1_CH][C1_K * C1_K + 1];
  float gs1[S1_WSZ + 1];
};

// Fused forward + backward-data, one 256-thread workgroup per image.
//
// Dataflow: thread t < 216 OWNS pool cell (o, pr, pc) — it computes the
// cell's 16 conv1 outputs (from the LDS image + staged weights), applies
// sigmoid, keeps them in REGISTERS, computes the pool output, and later
// runs the pool backward for the same 16 positions from those registers.
// That ownership removes the conv->pool and fc-bwd->pool-bwd barriers and
// all cross-thread a1 LDS traffic: the kernel has 3 barriers total
// ({stage} {conv+pool} {fc+residual} {fc-bwd + pool-bwd + loss}).
template <typename act_t, int MODE>
__global__ __launch_bounds__(256) void k_fwdbwd(
    const act_t* __restrict__ x, const float* __restrict__ params,
    act_t* __restrict__ a1g, act_t* __restrict__ a2g, float* __restrict__ yg,
    float* __restrict__ dzg, float* __restrict__ dz2g,
    act_t* __restrict__ dz1g, const int* __restrict__ labels,
    float* __restrict__ loss_accum, int* __restrict__ correct_accum, int B,
    int pool_mode, int loss_mode, float* __restrict__ grads,
    int wgrad_fuse) {
  __shared__ FwdLds L;
  const int b = blockIdx.x;
  if (b >= B) return;
  const int tid = threadIdx.x;

  // No staging phase: each thread loads its own 8x8 input window straight
  // from global (overlapping windows hit L1; one 1.5 KB image per block)
  // and the 173 conv/pool parameters broadcast through L1 — the kernel has
  // TWO barriers ({conv+pool} {fc+residual} {bwd}).
  if (MODE == MODE_TRAIN && wgrad_fuse) {
    if (tid < C1_CH * (C1_K * C1_K + 1))
      L.gw[tid / (C1_K * C1_K + 1)][tid % (C1_K * C1_K + 1)] = 0.f;
    else if (tid < C1_CH * (C1_K * C1_K + 1) + S1_WSZ + 1)
      L.gs1[tid - C1_CH * (C1_K * C1_K + 1)] = 0.f;
  }
  const act_t* xb = x + (size_t)b * IN_PIX;

  // ---- phase 1: conv1 + sigmoid + pool + sigmoid (no barrier between) ----
  const int o = tid / S1_PIX;
  const int pq = tid - o * S1_PIX;
  const int pr = pq / S1_W;
  const int pc = pq - pr * S1_W;
  float a1v[S1_K * S1_K];  // this cell's conv activations, kept live to bwd
  float a2v = 0.f;
  if (tid < S1_OUT) {
    // load the cell's 8x8 input window into registers (2 x 8B loads/row)
    float xw[8][8];
#pragma unroll
    for (int u = 0; u < 8; ++u)
#pragma unroll
      for (int v4 = 0; v4 < 2; ++v4)
        ld4f(xb + (pr * S1_K + u) * IN_W + pc * S1_K + v4 * 4,
             &xw[u][v4 * 4]);
    const float* w = params + OFF_C1W + o * C1_K * C1_K;
    const float cb = params[OFF_C1B + o];
    // pool preact: trainable weighted sum (reference) or max
    float pacc = pool_mode == 1 ? -1e30f : params[OFF_S1B];
#pragma unroll
    for (int i = 0; i < S1_K; ++i) {
#pragma unroll
      for (int j = 0; j < S1_K; ++j) {
        float acc = cb;
#pragma unroll
        for (int u = 0; u < C1_K; ++u)
#pragma unroll
          for (int v = 0; v < C1_K; ++v)
            acc += w[u * C1_K + v] * xw[i + u][j + v];
        const float av = sigmoidf_dev(acc);
        a1v[i * S1_K + j] = av;
        if (pool_mode == 1)
          pacc = fmaxf(pacc, av);
        else
          pacc += params[OFF_S1W + i * S1_K + j] * av;
      }
    }
    if (MODE == MODE_TRAIN) {
      // 4-wide activation stores per conv row
#pragma unroll
      for (int i = 0; i < S1_K; ++i)
        st4f(a1g + (size_t)b * C1_OUT + o * C1_PIX +
                 (pr * S1_K + i) * C1_W + pc * S1_K,
             &a1v[i * S1_K]);
    }
    a2v = sigmoidf_dev(pacc);
    L.a2s[tid] = a2v;
    if (MODE == MODE_TRAIN) stf(a2g + (size_t)b * S1_OUT + tid, a2v);
  }
  __syncthreads();

  // ---- phase 2: fc (216 -> 10) + sigmoid [+ loss residual] ----
  // 16 lanes per output class, 4-step shuffle tree; the group leader
  // applies bias+sigmoid and computes the residual in-phase.
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
      const float z = p + params[OFF_FB + k];
      if (loss_mode == 1) {
        L.ys[k] = z;  // raw logit; softmax finished below
      } else {
        const float v = sigmoidf_dev(z);
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
  }
  __syncthreads();
  if (loss_mode == 1) {
    // softmax over the 10 logits + CE residual (dz = onehot - softmax)
    if (tid == 0) {
      float mx = L.ys[0];
#pragma unroll
      for (int k = 1; k < FC_OUT; ++k) mx = fmaxf(mx, L.ys[k]);
      float sum = 0.f;
      float e[FC_OUT];
#pragma unroll
      for (int k = 0; k < FC_OUT; ++k) {
        e[k] = __expf(L.ys[k] - mx);
        sum += e[k];
      }
#pragma unroll
      for (int k = 0; k < FC_OUT; ++k) {
        const float v = e[k] / sum;
        L.ys[k] = v;
        if (yg != nullptr) yg[(size_t)b * FC_OUT + k] = v;
        if (MODE == MODE_TRAIN) {
          const float d = (k == labels[b] ? 1.0f : 0.0f) - v;
          L.dzs[k] = d;
          dzg[(size_t)b * FC_OUT + k] = d;
        }
      }
      if (MODE == MODE_TRAIN)
        L.sq[0] = -__logf(fmaxf(L.ys[labels[b]], 1e-30f));
    }
    __syncthreads();
  }

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

  // ---- phase 3: fc bwd -> pool preact grad -> pool bwd (no barriers) ----
  if (tid < S1_OUT) {
    float da = 0.f;
#pragma unroll
    for (int k = 0; k < FC_OUT; ++k)
      da += params[OFF_FW + k * FC_IN + tid] * L.dzs[k];
    const float d2 = da * a2v * (1.0f - a2v);
    dz2g[(size_t)b * S1_OUT + tid] = d2;
    // pool backward for this thread's own 16 conv positions (registers):
    // trainable -> d2 * kernel weight everywhere; max -> d2 at the argmax
    int best = 0;
    if (pool_mode == 1) {
      float bv = a1v[0];
#pragma unroll
      for (int t = 1; t < S1_K * S1_K; ++t)
        if (a1v[t] > bv) {
          bv = a1v[t];
          best = t;
        }
    }
    float dz1v[S1_K * S1_K];
#pragma unroll
    for (int i = 0; i < S1_K; ++i) {
#pragma unroll
      for (int j = 0; j < S1_K; ++j) {
        const float av = a1v[i * S1_K + j];
        const float dd =
            pool_mode == 1
                ? (i * S1_K + j == best ? d2 : 0.f)
                : d2 * params[OFF_S1W + i * S1_K + j];
        dz1v[i * S1_K + j] = dd * av * (1.0f - av);
      }
      st4f(dz1g + (size_t)b * C1_OUT + o * C1_PIX + (pr * S1_K + i) * C1_W +
               pc * S1_K,
           &dz1v[i * S1_K]);
    }
    if (wgrad_fuse) {
      // conv1 wgrad: this cell's 16 dz1 x its 8x8 input window (from LDS);
      // per-thread register accumulation, LDS-atomic combine per channel,
      // one global atomic per weight per block (in the final phase below).
      float cacc[C1_K * C1_K];
#pragma unroll
      for (int w2 = 0; w2 < C1_K * C1_K; ++w2) cacc[w2] = 0.f;
      float csum = 0.f;
#pragma unroll
      for (int t = 0; t < S1_K * S1_K; ++t) {
        const float d = dz1v[t];
        csum += d;
        const int r = pr * S1_K + t / S1_K;
        const int c = pc * S1_K + (t & 3);
#pragma unroll
        for (int u = 0; u < C1_K; ++u)
#pragma unroll
          for (int v = 0; v < C1_K; ++v)
            cacc[u * C1_K + v] += d * ldf(xb + (r + u) * IN_W + (c + v));
      }
#pragma unroll
      for (int w2 = 0; w2 < C1_K * C1_K; ++w2)
        atomicAdd(&L.gw[o][w2], cacc[w2]);
      atomicAdd(&L.gw[o][C1_K * C1_K], csum);
      if (pool_mode == 0) {
        // pool wgrad: dz2(own cell) x own a1 activations
#pragma unroll
        for (int t = 0; t < S1_K * S1_K; ++t)
          atomicAdd(&L.gs1[t], d2 * a1v[t]);
        atomicAdd(&L.gs1[S1_WSZ], d2);
      }
    }
  } else if (tid == S1_OUT && loss_accum != nullptr) {
    if (loss_mode == 1) {
      unsafeAtomicAdd(loss_accum, L.sq[0]);
    } else {
      float ssum = 0.f;
#pragma unroll
      for (int k = 0; k < FC_OUT; ++k) ssum += L.sq[k];
      unsafeAtomicAdd(loss_accum, sqrtf(ssum));
    }
  }
  if (MODE == MODE_TRAIN && wgrad_fuse) {
    __syncthreads();
    // conv grads carry the reference 1/(24*24) factor; pool bias /216
    constexpr float inv_pix = 1.0f / (float)C1_PIX;
    if (tid < C1_CH * (C1_K * C1_K + 1)) {
      const int o2 = tid / (C1_K * C1_K + 1);
      const int w2 = tid % (C1_K * C1_K + 1);
      const float v = L.gw[o2][w2] * inv_pix;
      if (w2 < C1_K * C1_K)
        unsafeAtomicAdd(&grads[OFF_C1W + o2 * C1_K * C1_K + w2], v);
      else
        unsafeAtomicAdd(&grads[OFF_C1B + o2], v);
    } else if (pool_mode == 0 &&
               tid < C1_CH * (C1_K * C1_K + 1) + S1_WSZ + 1) {
      const int w2 = tid - C1_CH * (C1_K * C1_K + 1);
      if (w2 < S1_WSZ)
        unsafeAtomicAdd(&grads[OFF_S1W + w2], L.gs1[w2]);
      else
        unsafeAtomicAdd(&grads[OFF_S1B], L.gs1[S1_WSZ] / (float)S1_OUT);
    }
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
    const float* __restrict__ dz2g, const act_t* __restrict__ dz1g,
    float* __restrict__ grads, int B, int GC, int GS, int FS, int roles) {
  const int tid = threadIdx.x;
  const int blk = blockIdx.x;
  const int lane = tid & 63;

  if (blk < C1_CH * GC) {
    if (!(roles & 1)) return;
    // ---- conv1: dW[o,i,j] = sum_{b,r,c} dz1[b,o,r,c] * x[b,r+i,c+j] / 576
    // Work item = 4 consecutive output columns of one row: 5 vectorized
    // 8-wide x row loads + one float4 dz1 load replace 104 scalar loads
    // (the v2 scalar form was address-issue-bound: profiles/ r1).  Two
    // groups are processed per loop iteration so the second group's loads
    // issue before the first group's FMAs — one memory-latency stall per
    // TWO groups instead of one per group.
    const int o = blk / GC;
    const int slice = blk - o * GC;
    constexpr int GROUPS_PER_ROW = C1_W / 4;  // 6
    float acc[C1_K * C1_K];
#pragma unroll
    for (int w = 0; w < C1_K * C1_K; ++w) acc[w] = 0.f;
    float bacc = 0.f;
    const int N = B * C1_H * GROUPS_PER_ROW;
    const int stride = GC * 256;
    for (int it = slice * 256 + tid; it < N; it += 2 * stride) {
      const int it2 = it + stride;
      float d4a[4], xra[C1_K][8];
      float d4b[4], xrb[C1_K][8];
      {
        const int bi = it / (C1_H * GROUPS_PER_ROW);
        const int rg = it - bi * (C1_H * GROUPS_PER_ROW);
        const int r = rg / GROUPS_PER_ROW;
        const int c0 = (rg - r * GROUPS_PER_ROW) * 4;
        ld4f(dz1g + (size_t)bi * C1_OUT + o * C1_PIX + r * C1_W + c0, d4a);
        const act_t* xb = x + (size_t)bi * IN_PIX + r * IN_W + c0;
#pragma unroll
        for (int i = 0; i < C1_K; ++i) ld8f(xb + i * IN_W, xra[i]);
      }
      if (it2 < N) {
        const int bi = it2 / (C1_H * GROUPS_PER_ROW);
        const int rg = it2 - bi * (C1_H * GROUPS_PER_ROW);
        const int r = rg / GROUPS_PER_ROW;
        const int c0 = (rg - r * GROUPS_PER_ROW) * 4;
        ld4f(dz1g + (size_t)bi * C1_OUT + o * C1_PIX + r * C1_W + c0, d4b);
        const act_t* xb = x + (size_t)bi * IN_PIX + r * IN_W + c0;
#pragma unroll
        for (int i = 0; i < C1_K; ++i) ld8f(xb + i * IN_W, xrb[i]);
      }
      bacc += d4a[0] + d4a[1] + d4a[2] + d4a[3];
#pragma unroll
      for (int i = 0; i < C1_K; ++i)
#pragma unroll
        for (int pp = 0; pp < 4; ++pp)
#pragma unroll
          for (int j = 0; j < C1_K; ++j)
            acc[i * C1_K + j] += d4a[pp] * xra[i][pp + j];
      if (it2 < N) {
        bacc += d4b[0] + d4b[1] + d4b[2] + d4b[3];
#pragma unroll
        for (int i = 0; i < C1_K; ++i)
#pragma unroll
          for (int pp = 0; pp < 4; ++pp)
#pragma unroll
            for (int j = 0; j < C1_K; ++j)
              acc[i * C1_K + j] += d4b[pp] * xrb[i][pp + j];
      }
    }
    // cross-wave pre-reduce in LDS: one hardware atomic per weight per
    // BLOCK (was one per wave -> 4x the same-address atomic traffic).
    __shared__ float red[4][C1_K * C1_K + 1];
    const int wv = tid >> 6;
#pragma unroll
    for (int w = 0; w < C1_K * C1_K; ++w) {
      const float v = wave_sum(acc[w]);
      if (lane == 0) red[wv][w] = v;
    }
    {
      const float v = wave_sum(bacc);
      if (lane == 0) red[wv][C1_K * C1_K] = v;
    }
    __syncthreads();
    constexpr float inv_pix = 1.0f / (float)C1_PIX;
    if (tid < C1_K * C1_K) {
      const float v =
          red[0][tid] + red[1][tid] + red[2][tid] + red[3][tid];
      unsafeAtomicAdd(&grads[OFF_C1W + o * C1_K * C1_K + tid], v * inv_pix);
    } else if (tid == C1_K * C1_K) {
      const float v = red[0][tid] + red[1][tid] + red[2][tid] + red[3][tid];
      unsafeAtomicAdd(&grads[OFF_C1B + o], v * inv_pix);
    }
  } else if (blk < C1_CH * GC + GS) {
    if (!(roles & 2)) return;
    // ---- pool: dW[i,j] = sum_{b,o,p,q} dz2[b,o,p,q] * a1[b,o,4p+i,4q+j]
    const int slice = blk - C1_CH * GC;
    float acc[S1_WSZ];
#pragma unroll
    for (int w = 0; w < S1_WSZ; ++w) acc[w] = 0.f;
    float bacc = 0.f;
    const int N = B * S1_OUT;
    const int stride = GS * 256;
    for (int it = slice * 256 + tid; it < N; it += 2 * stride) {
      const int it2 = it + stride;
      float da = 0.f, db = 0.f, ara[S1_K][S1_K], arb[S1_K][S1_K];
      {
        const int bi = it / S1_OUT;
        const int opq = it - bi * S1_OUT;
        const int oo = opq / S1_PIX;
        const int pq = opq - oo * S1_PIX;
        const int prr = pq / S1_W;
        const int pcc = pq - prr * S1_W;
        da = dz2g[it];
        const act_t* base = a1g + (size_t)bi * C1_OUT + oo * C1_PIX +
                            prr * S1_K * C1_W + pcc * S1_K;
#pragma unroll
        for (int i = 0; i < S1_K; ++i) ld4f(base + i * C1_W, ara[i]);
      }
      if (it2 < N) {
        const int bi = it2 / S1_OUT;
        const int opq = it2 - bi * S1_OUT;
        const int oo = opq / S1_PIX;
        const int pq = opq - oo * S1_PIX;
        const int prr = pq / S1_W;
        const int pcc = pq - prr * S1_W;
        db = dz2g[it2];
        const act_t* base = a1g + (size_t)bi * C1_OUT + oo * C1_PIX +
                            prr * S1_K * C1_W + pcc * S1_K;
#pragma unroll
        for (int i = 0; i < S1_K; ++i) ld4f(base + i * C1_W, arb[i]);
      }
      bacc += da;
#pragma unroll
      for (int i = 0; i < S1_K; ++i)
#pragma unroll
        for (int j = 0; j < S1_K; ++j) acc[i * S1_K + j] += da * ara[i][j];
      if (it2 < N) {
        bacc += db;
#pragma unroll
        for (int i = 0; i < S1_K; ++i)
#pragma unroll
          for (int j = 0; j < S1_K; ++j) acc[i * S1_K + j] += db * arb[i][j];
      }
    }
    __shared__ float red[4][S1_WSZ + 1];
    const int wv = tid >> 6;
#pragma unroll
    for (int w = 0; w < S1_WSZ; ++w) {
      const float v = wave_sum(acc[w]);
      if (lane == 0) red[wv][w] = v;
    }
    {
      const float v = wave_sum(bacc);
      if (lane == 0) red[wv][S1_WSZ] = v;
    }
    __syncthreads();
    if (tid < S1_WSZ) {
      const float v = red[0][tid] + red[1][tid] + red[2][tid] + red[3][tid];
      unsafeAtomicAdd(&grads[OFF_S1W + tid], v);
    } else if (tid == S1_WSZ) {
      const float v = red[0][tid] + red[1][tid] + red[2][tid] + red[3][tid];
      unsafeAtomicAdd(&grads[OFF_S1B], v / (float)S1_OUT);
    }
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
  // float4 per thread (the bucket is 16B-aligned); scalar tail
  const int i4 = blockIdx.x * 256 + threadIdx.x;
  const int i = i4 * 4;
  if (i + 3 < N_PARAMS) {
    float4 pv = *reinterpret_cast<float4*>(params + i);
    const float4 gv = *reinterpret_cast<float4*>(grads + i);
    pv.x += step * gv.x;
    pv.y += step * gv.y;
    pv.z += step * gv.z;
    pv.w += step * gv.w;
    *reinterpret_cast<float4*>(params + i) = pv;
    *reinterpret_cast<float4*>(grads + i) = make_float4(0.f, 0.f, 0.f, 0.f);
  } else if (i < N_PARAMS) {
    for (int u = i; u < N_PARAMS; ++u) {
      params[u] += step * grads[u];
      grads[u] = 0.f;
    }
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
                       float* y, float* dz, float* dz2, void* dz1,
                       const int* labels, float* loss_accum, int* correct,
                       int B, int act_is_bf16, int pool_mode, int loss_mode,
                       float* grads, int wgrad_fuse, hipStream_t stream) {
  dim3 grid(B), block(256);
  if (act_is_bf16 == 1) {
    hipLaunchKernelGGL((k_fwdbwd<bf16, MODE>), grid, block, 0, stream,
                       (const bf16*)x, params, (bf16*)a1, (bf16*)a2, y, dz,
                       dz2, (bf16*)dz1, labels, loss_accum, correct, B,
                       pool_mode, loss_mode, grads, wgrad_fuse);
  } else if (act_is_bf16 == 2) {
    hipLaunchKernelGGL((k_fwdbwd<fp16, MODE>), grid, block, 0, stream,
                       (const fp16*)x, params, (fp16*)a1, (fp16*)a2, y, dz,
                       dz2, (fp16*)dz1, labels, loss_accum, correct, B,
                       pool_mode, loss_mode, grads, wgrad_fuse);
  } else {
    hipLaunchKernelGGL((k_fwdbwd<float, MODE>), grid, block, 0, stream,
                       (const float*)x, params, (float*)a1, (float*)a2, y, dz,
                       dz2, (float*)dz1, labels, loss_accum, correct, B,
                       pool_mode, loss_mode, grads, wgrad_fuse);
  }
  return (int)hipGetLastError();
}
}  // namespace

extern "C" {

int pcnn_launch_fwdbwd_ex2(const void* x, const float* params, void* a1,
                           void* a2, float* y, float* dz, float* dz2,
                           void* dz1, const int* labels, float* loss_accum,
                           int* correct, int B, int act_is_bf16, int mode,
                           int pool_mode, int loss_mode, float* grads,
                           int wgrad_fuse, void* stream) {
  hipStream_t s = (hipStream_t)stream;
  switch (mode) {
    case MODE_TRAIN:
      return launch_fwdbwd_mode<MODE_TRAIN>(x, params, a1, a2, y, dz, dz2,
                                            dz1, labels, loss_accum, correct,
                                            B, act_is_bf16, pool_mode,
                                            loss_mode, grads, wgrad_fuse, s);
    case MODE_EVAL:
      return launch_fwdbwd_mode<MODE_EVAL>(x, params, a1, a2, y, dz, dz2, dz1,
                                           labels, loss_accum, correct, B,
                                           act_is_bf16, pool_mode, loss_mode,
                                           grads, wgrad_fuse, s);
    default:
      return launch_fwdbwd_mode<MODE_INFER>(x, params, a1, a2, y, dz, dz2,
                                            dz1, labels, loss_accum, correct,
                                            B, act_is_bf16, pool_mode,
                                            loss_mode, grads, wgrad_fuse, s);
  }
}

int pcnn_launch_fwdbwd_ex(const void* x, const float* params, void* a1,
                          void* a2, float* y, float* dz, float* dz2,
                          void* dz1, const int* labels, float* loss_accum,
                          int* correct, int B, int act_is_bf16, int mode,
                          int pool_mode, int loss_mode, void* stream) {
  return pcnn_launch_fwdbwd_ex2(x, params, a1, a2, y, dz, dz2, dz1, labels,
                                loss_accum, correct, B, act_is_bf16, mode,
                                pool_mode, loss_mode, nullptr, 0, stream);
}

int pcnn_launch_fwdbwd(const void* x, const float* params, void* a1, void* a2,
                       float* y, float* dz, float* dz2, void* dz1,
                       const int* labels, float* loss_accum, int* correct,
                       int B, int act_is_bf16, int mode, void* stream) {
  return pcnn_launch_fwdbwd_ex(x, params, a1, a2, y, dz, dz2, dz1, labels,
                               loss_accum, correct, B, act_is_bf16, mode, 0,
                               0, stream);
}

// chunk_imgs is the conv1 slices-per-channel knob (GC); <=0 -> default.
int pcnn_launch_wgrad_ex(const void* x, const void* a1, const void* a2,
                         const float* dz, const float* dz2, const void* dz1,
                         float* grads, int B, int act_is_bf16, int chunk_imgs,
                         int roles, void* stream) {
  // Batch-adaptive defaults: ~36 (image,position) items per conv thread,
  // pool slices at GC/4, fc batch slices of ~64 images.
  int GC = chunk_imgs > 0 ? chunk_imgs : (int)(4.0f * __builtin_cbrtf((float)B) + 0.5f);
  if (GC < 2) GC = 2;
  if (GC > 256) GC = 256;
  if (roles == 4) GC = 0;  // fc-only launch (fused-wgrad mode)
  // pool-role slices: ~10 (image,cell) items per thread (2 blocks at B=64
  // left the pool role as the kernel's longest pole)
  int GS = (B * S1_OUT + 256 * 5 - 1) / (256 * 5);
  if (GS < 2) GS = 2;
  if (GS > 96) GS = 96;
  if (roles == 4) GS = 0;
  int FS = B / 64;
  if (FS < 1) FS = 1;
  if (FS > 32) FS = 32;
  dim3 grid(C1_CH * GC + GS + FS * FC_BLOCKS), block(256);
  hipStream_t s = (hipStream_t)stream;
  if (act_is_bf16 == 1) {
    hipLaunchKernelGGL((k_wgrad<bf16>), grid, block, 0, s, (const bf16*)x,
                       (const bf16*)a1, (const bf16*)a2, dz, dz2,
                       (const bf16*)dz1, grads, B, GC, GS, FS, roles);
  } else if (act_is_bf16 == 2) {
    hipLaunchKernelGGL((k_wgrad<fp16>), grid, block, 0, s, (const fp16*)x,
                       (const fp16*)a1, (const fp16*)a2, dz, dz2,
                       (const fp16*)dz1, grads, B, GC, GS, FS, roles);
  } else {
    hipLaunchKernelGGL((k_wgrad<float>), grid, block, 0, s, (const float*)x,
                       (const float*)a1, (const float*)a2, dz, dz2,
                       (const float*)dz1, grads, B, GC, GS, FS, roles);
  }
  return (int)hipGetLastError();
}

int pcnn_launch_wgrad(const void* x, const void* a1, const void* a2,
                      const float* dz, const float* dz2, const void* dz1,
                      float* grads, int B, int act_is_bf16, int chunk_imgs,
                      void* stream) {
  return pcnn_launch_wgrad_ex(x, a1, a2, dz, dz2, dz1, grads, B, act_is_bf16,
                              chunk_imgs, 7, stream);
}

int pcnn_launch_update(float* params, float* grads, float step, void* stream) {
  dim3 grid((N_PARAMS / 4 + 255) / 256), block(256);
  hipLaunchKernelGGL(k_update, grid, block, 0, (hipStream_t)stream, params,
                     grads, step);
  return (int)hipGetLastError();
}

const char* pcnn_hip_error_string(int err) {
  return hipGetErrorString((hipError_t)err);
}
}
