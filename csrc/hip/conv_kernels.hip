// General conv-net kernels for gfx950 (CDNA4 / MI355X): the im2col + MFMA
// GEMM path (BASELINE.json config #4: 3x(conv5x5+pool)+FC on 32x32x3).
//
// Layout conventions (chosen MFMA-first):
//   * Activations are NHWC:  a[B][H][W][C]  (act_t = bf16 / fp16 / fp32);
//     the GEMM view of a conv output is C[M][N] with M = B*OH*OW rows in
//     (b, oh, ow) order and N = Cout — i.e. exactly the NHWC tensor.
//   * im2col columns:  cols[M][KcP] with kc = (i*K + j)*Cin + ci and KcP =
//     Kc rounded up to 32 (zero-padded) so every GEMM K-loop is whole MFMA
//     steps.  Consecutive kc == consecutive input channels == contiguous
//     NHWC memory, so im2col reads and GEMM stages are coalesced.
//   * Conv weights are stored [KcP][Cout] fp32 (master); pad rows are zero
//     and stay zero (their gradients are identically zero).  Kernels cast
//     fp32 -> bf16 while staging into LDS, so no separate cast pass exists.
//
// GEMM kernels use __builtin_amdgcn_mfma_f32_16x16x32_bf16 (gfx950 2xK
// form), 64x64 C-tiles, BK=32, 4 waves x 4 fragments, single-buffered LDS
// with +8 bf16 row padding (conflict-free ds_read_b128 fragment reads).
// Correctness-first structure (the "step-0/1" shape of the CDNA GEMM
// ladder); these GEMMs are microseconds at our sizes.
//
// Numerics: all accumulation fp32 (MFMA accumulators); activations and the
// backward-data tensors stored act_t; parameters/gradients fp32.

#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <hip/hip_runtime.h>

#include <type_traits>

#include "../lenet_dims.h"

namespace pcnn_deep {

using bf16 = __hip_bfloat16;
using fp16 = __half;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(4))) unsigned int u32x4;

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

// 8 consecutive elements -> fp32 (16-byte-aligned bf16/fp16, 32B fp32).
__device__ __forceinline__ void ld8v(const bf16* p, float* out) {
  const ushort4 a = *reinterpret_cast<const ushort4*>(p);
  const ushort4 b = *reinterpret_cast<const ushort4*>(p + 4);
  const bf16* ea = reinterpret_cast<const bf16*>(&a);
  const bf16* eb = reinterpret_cast<const bf16*>(&b);
#pragma unroll
  for (int u = 0; u < 4; ++u) {
    out[u] = (float)ea[u];
    out[u + 4] = (float)eb[u];
  }
}
__device__ __forceinline__ void ld8v(const fp16* p, float* out) {
  const ushort4 a = *reinterpret_cast<const ushort4*>(p);
  const ushort4 b = *reinterpret_cast<const ushort4*>(p + 4);
  const fp16* ea = reinterpret_cast<const fp16*>(&a);
  const fp16* eb = reinterpret_cast<const fp16*>(&b);
#pragma unroll
  for (int u = 0; u < 4; ++u) {
    out[u] = (float)ea[u];
    out[u + 4] = (float)eb[u];
  }
}
__device__ __forceinline__ void ld8v(const float* p, float* out) {
  const float4 a = *reinterpret_cast<const float4*>(p);
  const float4 b = *reinterpret_cast<const float4*>(p + 4);
  out[0] = a.x; out[1] = a.y; out[2] = a.z; out[3] = a.w;
  out[4] = b.x; out[5] = b.y; out[6] = b.z; out[7] = b.w;
}

// Unaligned 8-element load -> fp32 (element-aligned only — the small-Cin
// im2col spans start at arbitrary channel offsets).  The aligned(2/4)
// vector types let LLVM emit unaligned-capable global loads (or split
// them) instead of UB.
__device__ __forceinline__ void ld8v_u(const bf16* p, float* out) {
  bf16 t[8];
  __builtin_memcpy(t, p, 16);
#pragma unroll
  for (int u = 0; u < 8; ++u) out[u] = (float)t[u];
}
__device__ __forceinline__ void ld8v_u(const fp16* p, float* out) {
  fp16 t[8];
  __builtin_memcpy(t, p, 16);
#pragma unroll
  for (int u = 0; u < 8; ++u) out[u] = (float)t[u];
}
__device__ __forceinline__ void ld8v_u(const float* p, float* out) {
  __builtin_memcpy(out, p, 32);
}

// Magic-multiply unsigned division (divisor known on the host): exact
// for n < 2^38/d and n*M < 2^64, i.e. n < 2^26 with d <= 4096 — the
// launcher checks the domain and falls back to the 64-bit-div kernel
// variant beyond it.  A runtime u32 divide is ~25 VALU ops (and a 64-bit
// one a ~100-op libcall); this is a 32x32->64 mul + shift.
constexpr int FDIV_SH = 38;
__host__ __device__ __forceinline__ unsigned long long fdiv_magic(
    unsigned d) {
  return ((1ULL << FDIV_SH) + d - 1) / d;
}
__device__ __forceinline__ unsigned fdiv(unsigned n, unsigned long long M) {
  return (unsigned)((n * M) >> FDIV_SH);
}

// ---------------------------------------------------------------------------
// im2col (NHWC, same-padding):
//   cols[(b*OH+oh)*OW+ow][ (i*K+j)*Cin+ci ] = x[b][oh+i-P][ow+j-P][ci] or 0
// One thread per cols element; kc is the fast axis (coalesced stores, and
// coalesced loads since ci is the fast axis of NHWC x).
// ---------------------------------------------------------------------------
// Span-walk formulation: one thread owns FOUR consecutive 8-element
// spans (32 elements, KcP % 32 == 0 by construction) of one cols row.
// The (b, oh, ow) decode and the kc -> (i, j, ci) decode happen ONCE;
// the walk across spans is incremental (no divisions).  For Cin % 8 == 0
// a span always lies inside one input pixel (ci % 8 == 0 and 8 <= Cin),
// so it is a single 16B load; smaller Cin walks elementwise with
// incremental counters (the per-element div/mod version was 83% VALUBusy
// on index math).
template <typename act_t, bool FAST>
__global__ void k_im2col(const act_t* __restrict__ x, act_t* __restrict__ cols,
                         int B, int H, int W, int Cin, int K, int P,
                         int KcP, unsigned long long fd_tpr,
                         unsigned long long fd_w, unsigned long long fd_h) {
  const long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long total = ((long long)B * H * W * KcP) / 32;
  if (idx >= total) return;
  int kc, ow, oh, b;
  long long m;
  if (FAST) {
    const unsigned n = (unsigned)idx;
    const unsigned tpr = (unsigned)(KcP / 32);  // threads per cols row
    const unsigned mq = fdiv(n, fd_tpr);
    kc = (int)(n - mq * tpr) * 32;
    const unsigned bhq = fdiv(mq, fd_w);
    ow = (int)(mq - bhq * (unsigned)W);
    const unsigned bq = fdiv(bhq, fd_h);
    oh = (int)(bhq - bq * (unsigned)H);
    b = (int)bq;
    m = mq;
  } else {
    kc = (int)((idx * 32) % KcP);
    m = (idx * 32) / KcP;
    ow = (int)(m % W);
    const long long bh = m / W;
    oh = (int)(bh % H);
    b = (int)(bh / H);
  }
  const int Kc = K * K * Cin;
  const int rowc = K * Cin;
  const act_t* __restrict__ xb = x + (long long)b * H * W * Cin;
  act_t* __restrict__ orow = cols + m * KcP;

  // span-0 decode — the only divisions in the kernel
  int i = kc / rowc;
  int t = kc - i * rowc;
  int j = t / Cin;
  int ci = t - j * Cin;
  const bool vec = (Cin % 8) == 0;

  for (int s = 0; s < 4; ++s, kc += 8) {
    act_t out[8];
    if (kc >= Kc) {
#pragma unroll
      for (int u = 0; u < 8; ++u) out[u] = (act_t)0.f;
    } else if (vec) {
      const int ih = oh + i - P;
      const int iw = ow + j - P;
      if (ih >= 0 && ih < H && iw >= 0 && iw < W) {
        float v8[8];
        ld8v(xb + ((long long)ih * W + iw) * Cin + ci, v8);
#pragma unroll
        for (int u = 0; u < 8; ++u) out[u] = (act_t)v8[u];
      } else {
#pragma unroll
        for (int u = 0; u < 8; ++u) out[u] = (act_t)0.f;
      }
      ci += 8;
      if (ci >= Cin) {
        ci = 0;
        if (++j >= K) {
          j = 0;
          ++i;
        }
      }
    } else {
      // Small Cin: consecutive kc within one KERNEL ROW map to
      // consecutive x memory (kc = (i*K+j)*Cin+ci, ci fastest, and
      // NHWC x is (..., iw, ci)-contiguous) — so an 8-span that stays
      // inside the row AND inside the image is ONE 16B load.  Border /
      // row-crossing spans use clamped unconditional scalar loads +
      // select-to-zero (a branchy guarded chain was this kernel's
      // whole cost on the Cin=3 stage).
      const int r0 = kc - (i * K + j) * Cin + j * Cin;  // = kc % rowc
      const int jlast = j + (ci + 7) / Cin;
      const int ih = oh + i - P;
      if (r0 + 8 <= rowc && kc + 8 <= Kc && ih >= 0 && ih < H &&
          ow + j - P >= 0 && ow + jlast - P < W) {
        float v8[8];
        ld8v_u(xb + ((long long)ih * W + (ow + j - P)) * Cin + ci, v8);
#pragma unroll
        for (int u = 0; u < 8; ++u) out[u] = (act_t)v8[u];
        // advance (i, j, ci) by 8 positions
        ci += 8;
        j += ci / Cin;
        ci %= Cin;
        if (j >= K) {
          i += j / K;
          j %= K;
        }
      } else {
#pragma unroll
        for (int u = 0; u < 8; ++u) {
          const int ih2 = oh + i - P;
          const int iw = ow + j - P;
          const int ihc = min(max(ih2, 0), H - 1);
          const int iwc = min(max(iw, 0), W - 1);
          const float val =
              ldf(xb + ((long long)ihc * W + iwc) * Cin + ci);
          out[u] = (act_t)((kc + u < Kc && ih2 == ihc && iw == iwc)
                               ? val
                               : 0.f);
          if (++ci >= Cin) {
            ci = 0;
            if (++j >= K) {
              j = 0;
              ++i;
            }
          }
        }
      }
    }
    if (sizeof(act_t) == 2)
      // NOT __builtin_nontemporal_store: cols is re-read immediately by
      // the GEMM and partially L2-resident — NT stores cost 15% end-to-end
      *reinterpret_cast<uint4*>(orow + kc) =
          *reinterpret_cast<const uint4*>(out);
    else
#pragma unroll
      for (int u = 0; u < 8; ++u) orow[kc + u] = out[u];
  }
}

// 8-wide implicit-im2col gather (GEMM A-operand staged straight from the
// NHWC activation, no materialized cols buffer).  Requires Cin % 8 == 0
// (launcher-enforced): an 8-span then always lies inside ONE input pixel,
// so it is a single clamped UNCONDITIONAL 16B load with a select-to-zero
// afterwards — the round-1 implicit path lost to the materialized one
// because its guarded gather de-pipelined the K-loop (hipcc branches
// around conditional loads); the clamp+select idiom is what fixed the
// materialized path and is applied here.  The (i, j, ci) decode is two
// magic-multiply divisions (fd_cin = fdiv_magic(Cin), fd_k =
// fdiv_magic(K)).  valid=false spans (image border, kc >= Kc pad, m past
// M) read a clamped in-bounds address and are zeroed after the load.
template <typename act_t>
__device__ __forceinline__ void im2col8f(const act_t* __restrict__ x, int H,
                                         int W, int Cin, int K, int P, int b,
                                         int oh, int ow, int kc0, int Kc,
                                         unsigned long long fd_cin,
                                         unsigned long long fd_k, bool valid,
                                         float* v8) {
  const unsigned p = fdiv((unsigned)kc0, fd_cin);
  const int ci = kc0 - (int)p * Cin;
  const unsigned pi = fdiv(p, fd_k);
  const int j = (int)(p - pi * (unsigned)K);
  const int i = (int)pi;
  const int ih = oh + i - P;
  const int iw = ow + j - P;
  const int ihc = min(max(ih, 0), H - 1);
  const int iwc = min(max(iw, 0), W - 1);
  ld8v(x + (((long long)b * H + ihc) * W + iwc) * Cin + ci, v8);
  if (!(valid && kc0 < Kc && ih == ihc && iw == iwc)) {
#pragma unroll
    for (int u = 0; u < 8; ++u) v8[u] = 0.f;
  }
}

// ---------------------------------------------------------------------------
// MFMA GEMM fragment maps for mfma_f32_16x16x32_bf16 (validated on-device
// by k_mfma_selftest / tests):
//   A[16x32]: lane l, reg r (0..7):  A[row = l&15][k = (l>>4)*8 + r]
//   B[32x16]: lane l, reg r:         B[k = (l>>4)*8 + r][col = l&15]
//   C[16x16]: lane l, reg r (0..3):  C[row = (l>>4)*4 + r][col = l&15]
// ---------------------------------------------------------------------------

constexpr int BM = 64, BN = 64, BK = 64;   // wgrad M-chunk per barrier pair
constexpr int BKC = 32;  // k_gemm K-step (measured: 128 halves the
                         // iteration count but costs 187 VGPRs -> occupancy
                         // 2 waves/SIMD and loses 15% end-to-end; 64 keeps
                         // 68 VGPRs / occupancy 5)

template <int TBM, int KD>
struct GemmLdsT {
  __bf16 As[TBM][KD + 8];  // A tile, [m][k]; +8 rows: conflict-free b128
  __bf16 Bs[BN][KD + 8];   // B tile, [n][k]  (transposed image: frag reads
                           // are contiguous along k for both operands)
};
using GemmLds = GemmLdsT<BM, BK>;

// Load an 8-element bf16 fragment from an LDS row.
__device__ __forceinline__ bf16x8 frag_from_lds(const __bf16* row, int k0) {
  return *reinterpret_cast<const bf16x8*>(row + k0);
}

// Generic C[M][N] = epilogue(A[M][ldA] @ B [K][N] + bias):
//   A: act_t, row-major, leading dim ldA (>= K, multiple of 32 used).
//   Bsrc: fp32.  b_kxn = true: Bsrc is [K][N] (stage-transposed to LDS);
//                 false: Bsrc is [N][K] (staged directly).
//   epilogue: 0 = plain store, 1 = bias + sigmoid,
//             2 = sigmoid-backward:  C = acc * epi * (1 - epi)   (epi is
//                 the sigmoid OUTPUT tensor, same [M][ldC] shape as C —
//                 used by the implicit dgrad-as-conv, which writes the
//                 previous stage's pool preact grad directly and kills
//                 the dcols round-trip + k_col2im_sigbwd pass),
//             3 = mode 1 PLUS the trainable pool forward fused in: the
//                 sigmoid activations are also staged to LDS and each WG
//                 emits its tile's pool outputs (pw = [PK*PK] kernel then
//                 scalar bias, pout = pooled sigmoid output).  Launcher-
//                 enforced preconditions: TBM == 64, one n-tile
//                 (N <= 64), 64 % XW == 0, (64/XW) % PK == 0,
//                 XH % PK == 0 — every 2x2 window then lies inside one
//                 M-tile for all DeepCNN shapes (XW in {8,16,32}).
//   C: act_t [M][ldC].
// Bpre (optional): pre-cast bf16 B in [N][K] row-per-output-column layout;
// when non-null it replaces Bsrc/b_kxn and stages with two 16B copies per
// thread per tile.
// imx != null: the A operand is the im2col view of NHWC imx (implicit
// GEMM — no materialized cols buffer); A/ldA are ignored, K = KcP, and
// the imx geometry is (XH, XW, XC, XK, XP) with XC % 8 == 0
// (launcher-enforced; fd_cin/fd_k are the magic divisors).
// TBM: M-tile (64 or 128).  At 128 each wave owns two 16-row fragments
// (16 MFMAs per K-step — double the compute per barrier pair) at +9 KB
// LDS; the launcher picks it for large-M calls.
// Split-K (c32 != null): a launch whose (mtiles x ntiles) grid cannot
// fill 256 CUs fully exposes the per-iteration stage/barrier stall (a
// 64-WG stage-2 GEMM measured 47 us vs 20 us for the same FLOPs at 1024
// WGs).  The grid gains a K-chunk axis: each WG covers tpc BKC-tiles of
// K and stores its fp32 partial tile to slab `kchunk` of c32
// [ks][M][N]; k_split_epi sums the slabs and applies the epilogue
// (deterministic — no atomics, so graph replay and tests stay bitwise
// reproducible).
template <typename act_t, int TBM>
__global__ __launch_bounds__(256) void k_gemm(
    const act_t* __restrict__ A, const float* __restrict__ Bsrc,
    const __bf16* __restrict__ Bpre, const float* __restrict__ bias,
    act_t* __restrict__ C, long long M, int K, int N, int ldA, int ldC,
    int b_kxn, int epilogue, const act_t* __restrict__ imx, int XH, int XW,
    int XC, int XK, int XP, const act_t* __restrict__ epi,
    unsigned long long fd_cin, unsigned long long fd_k,
    const float* __restrict__ pw, act_t* __restrict__ pout, int PK,
    float* __restrict__ c32, int tpc) {
  __shared__ GemmLdsT<TBM, BKC> Lb[2];  // double-buffered tiles
  constexpr int RF = TBM / 64;        // row fragments per wave
  constexpr int TPR = 256 / TBM;      // staging threads per A row
  constexpr int SPAN = BKC / TPR;     // k-span per staging thread
  constexpr int NCH = SPAN / 8;       // 8-chunks per staging thread
  constexpr int SPANB = BKC / 4;      // k-span per B staging thread
  constexpr int NCHB = SPANB / 8;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wv = tid >> 6;
  const int ntiles = (N + BN - 1) / BN;
  const long long mtiles = (M + TBM - 1) / TBM;
  const long long mn = mtiles * ntiles;
  const int kchunk = (int)(blockIdx.x / mn);
  const long long rest = blockIdx.x % mn;
  const long long mtile = rest / ntiles;
  const int ntile = (int)(rest % ntiles);
  const int k_lo = c32 ? kchunk * tpc * BKC : 0;
  const int k_hi = c32 ? min(K, k_lo + tpc * BKC) : K;
  const long long m0 = mtile * TBM;
  const int n0 = ntile * BN;
  const int nf = min(BN, N - n0) / 16;  // fragments along N (N % 16 == 0)

  f32x4 acc[RF][BN / 16];
#pragma unroll
  for (int rf = 0; rf < RF; ++rf)
#pragma unroll
    for (int f = 0; f < BN / 16; ++f) acc[rf][f] = {0.f, 0.f, 0.f, 0.f};

  const int row_a = tid / TPR;
  const int kq = (tid % TPR) * SPAN;
  const int bkq = (tid & 3) * SPANB;  // B staging: 4 threads per row
  // T14-style software pipeline: next tile's global loads are issued into
  // registers BEFORE the MFMA block; the LDS write happens after the read
  // barrier.
  float ra[NCH][8];
  float rb[(NCHB > 2 ? NCHB : 2)][8];  // the b_kxn staging needs 2 chunks
                                       // even when BKC < 64
  const long long m_a = m0 + row_a;
  // implicit-A: decode this thread's im2col row position once (from the
  // CLAMPED row — loads are unconditional, so the address must stay
  // in-bounds even for m_a >= M rows, which are zero-selected)
  int ib = 0, ioh = 0, iow = 0, iKc = 0;
  if (imx != nullptr) {
    const long long mm = m_a < M ? m_a : M - 1;
    iow = (int)(mm % XW);
    const long long bh = mm / XW;
    ioh = (int)(bh % XH);
    ib = (int)(bh / XH);
    iKc = XK * XK * XC;
  }

  auto load_regs = [&](int kt) {
#pragma unroll
    for (int h = 0; h < NCH; ++h) {
      const int kk = kq + h * 8;
      const bool ok = m_a < M && (kt + kk) < K;
      if (imx != nullptr) {
        im2col8f(imx, XH, XW, XC, XK, XP, ib, ioh, iow, kt + kk, iKc,
                 fd_cin, fd_k, ok, ra[h]);
      } else {
        // clamped unconditional load + select-to-zero: a load behind a
        // thread-varying guard de-pipelines the whole K-loop (hipcc
        // branches around it; CDNA4 lesson from the implicit-im2col
        // experiment).  K % 8 == 0 and K <= ldA keep the clamp in-bounds.
        const long long mm = m_a < M ? m_a : M - 1;
        const int kcl = (kt + kk) < K ? (kt + kk) : K - 8;
        ld8v(A + mm * ldA + kcl, ra[h]);
        if (!ok) {
#pragma unroll
          for (int u = 0; u < 8; ++u) ra[h][u] = 0.f;
        }
      }
    }
    if (Bpre != nullptr) {
      // bf16 [N][K] rows: two clamped unconditional vector loads per
      // thread, zeros selected in afterwards (see the A-path note)
      const int n = tid >> 2;
      const int nn = (n0 + n) < N ? (n0 + n) : N - 1;
#pragma unroll
      for (int h = 0; h < NCHB; ++h) {
        const int kk = bkq + h * 8;
        const int kcl = (kt + kk) < K ? (kt + kk) : K - 8;
        ld8v(reinterpret_cast<const bf16*>(Bpre + (long long)nn * K + kcl),
             rb[h]);
        if (!((n0 + n) < N && (kt + kk) < K)) {
#pragma unroll
          for (int u = 0; u < 8; ++u) rb[h][u] = 0.f;
        }
      }
    } else if (b_kxn) {
      // Bsrc[K][N]: read rows k (coalesced along n), write transposed;
      // each thread covers ceil(BKC/64) k-rows 64 apart (rows >= BKC
      // skipped when BKC < 64)
#pragma unroll
      for (int kb = 0; kb < (BKC + 63) / 64; ++kb) {
        const int k = (tid >> 2) + kb * 64;
        if (k >= BKC) continue;
        const int nq = (tid & 3) * 16;
        const bool kok = (kt + k) < K;
#pragma unroll
        for (int h = 0; h < 2; ++h)
#pragma unroll
          for (int u = 0; u < 8; ++u) {
            const int n = n0 + nq + h * 8 + u;
            rb[kb * 2 + h][u] =
                (kok && n < N) ? Bsrc[(long long)(kt + k) * N + n] : 0.f;
          }
      }
    } else {
      // Bsrc[N][K]: row per n
      const int n = tid >> 2;
      const bool ok = (n0 + n) < N;
#pragma unroll
      for (int h = 0; h < NCHB; ++h) {
        const int kk = bkq + h * 8;
        const float* src = Bsrc + (long long)(n0 + n) * K + kt + kk;
        const bool kok = ok && (kt + kk) < K;
#pragma unroll
        for (int u = 0; u < 8; ++u) rb[h][u] = kok ? src[u] : 0.f;
      }
    }
  };

  // pack 8 fp32 -> one 16B bf16 LDS store (the row-major A/B images are
  // contiguous along k and 16B aligned: kk % 8 == 0, row pitch 72*2 B)
  auto st8 = [](__bf16* dst, const float* v) {
    union {
      __bf16 h[8];
      uint4 u;
    } t;
#pragma unroll
    for (int u2 = 0; u2 < 8; ++u2) t.h[u2] = (__bf16)v[u2];
    *reinterpret_cast<uint4*>(dst) = t.u;
  };
  auto write_lds = [&](GemmLdsT<TBM, BKC>& L) {
#pragma unroll
    for (int h = 0; h < NCH; ++h) st8(&L.As[row_a][kq + h * 8], ra[h]);
    if (Bpre == nullptr && b_kxn) {
#pragma unroll
      for (int kb = 0; kb < (BKC + 63) / 64; ++kb) {
        const int k = (tid >> 2) + kb * 64;
        if (k >= BKC) continue;
        const int nq = (tid & 3) * 16;
#pragma unroll
        for (int h = 0; h < 2; ++h)
#pragma unroll
          for (int u = 0; u < 8; ++u)
            L.Bs[nq + h * 8 + u][k] = (__bf16)rb[kb * 2 + h][u];
      }
    } else {
      const int n = tid >> 2;
#pragma unroll
      for (int h = 0; h < NCHB; ++h) st8(&L.Bs[n][bkq + h * 8], rb[h]);
    }
  };

  // Double-buffered pipeline, ONE barrier per K-iteration: while the
  // MFMAs read tile t from Lb[p], tile t+1 (issued a full iteration ago)
  // is written to Lb[1-p] and tile t+2's loads are issued — the
  // s_waitcnt for a tile's global loads lands ~one iteration after
  // issue instead of right after the MFMA block.
  load_regs(k_lo);
  write_lds(Lb[0]);
  if (k_lo + BKC < k_hi) load_regs(k_lo + BKC);
  __syncthreads();              // Lb[0] visible
  int p = 0;
  for (int kt = k_lo; kt < k_hi; kt += BKC, p ^= 1) {
    auto& L = Lb[p];
    // wave wv owns C rows [wv*16*RF, +16*RF); BKC/32 32-deep MFMA sub-steps
#pragma unroll
    for (int kk = 0; kk < BKC / 32; ++kk) {
#pragma unroll
      for (int rf = 0; rf < RF; ++rf) {
        const bf16x8 a0 =
            frag_from_lds(L.As[(wv * RF + rf) * 16 + (lane & 15)],
                          kk * 32 + (lane >> 4) * 8);
#pragma unroll
        for (int f = 0; f < BN / 16; ++f) {
          if (f < nf) {
            const bf16x8 bf = frag_from_lds(L.Bs[f * 16 + (lane & 15)],
                                            kk * 32 + (lane >> 4) * 8);
            acc[rf][f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a0, bf, acc[rf][f], 0, 0, 0);
          }
        }
      }
    }
    if (kt + BKC < k_hi) {
      write_lds(Lb[p ^ 1]);  // waits on tile t+1's loads here
      if (kt + 2 * BKC < k_hi) load_regs(kt + 2 * BKC);
    }
    __syncthreads();  // reads of Lb[p] done AND Lb[1-p] complete
  }

  // split-K: store the fp32 partial tile to this chunk's slab and exit
  if (c32 != nullptr) {
    float* slab = c32 + (long long)kchunk * M * N;
#pragma unroll
    for (int rf = 0; rf < RF; ++rf) {
      const int crow = (wv * RF + rf) * 16 + ((tid & 63) >> 4) * 4;
#pragma unroll
      for (int f = 0; f < BN / 16; ++f) {
        if (f < nf) {
          const int n = n0 + f * 16 + (tid & 15);
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const long long m = m0 + crow + r;
            if (m < M && n < N) slab[m * N + n] = acc[rf][f][r];
          }
        }
      }
    }
    return;
  }

  // epilogue: lane l, reg r -> C[row=(l>>4)*4+r][col=l&15] of its fragment
  const int ccol = lane & 15;
  // mode 3 stages the sigmoid acts to LDS for the fused pool (reuse of
  // the now-idle tile buffers; [64][64] fp32 = 16 KB < sizeof Lb[0..1])
  float* const sc = reinterpret_cast<float*>(&Lb[0]);
#pragma unroll
  for (int rf = 0; rf < RF; ++rf) {
    const int crow = (wv * RF + rf) * 16 + (lane >> 4) * 4;
#pragma unroll
    for (int f = 0; f < BN / 16; ++f) {
      if (f < nf) {
        const int n = n0 + f * 16 + ccol;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const long long m = m0 + crow + r;
          if (m < M && n < N) {
            float v = acc[rf][f][r];
            if (epilogue >= 1 && epilogue != 2) {
              v = sigmoidf_dev(v + bias[n]);
            } else if (epilogue == 2) {
              const float e = ldf(epi + m * ldC + n);
              v = v * e * (1.0f - e);
            }
            stf(C + m * ldC + n, v);
            if (epilogue == 3) sc[(crow + r) * 64 + n] = v;
          }
        }
      }
    }
  }
  if (epilogue == 3 && TBM == 64) {
    __syncthreads();
    // this tile covers whole conv rows (64 % XW == 0) pool-aligned
    // ((64/XW) % PK == 0): emit its (64 / PK^2) * N pool outputs
    const int OW = XW / PK;
    const long long row0 = m0 / XW;       // first (b*XH + h) row
    const int npos = 64 / (PK * PK);      // pooled positions in the tile
    const float pb = pw[PK * PK];
    for (int o = tid; o < npos * N; o += 256) {
      const int lp = o / N;               // local pooled index
      const int n = o - lp * N;
      const int pr = lp / OW;             // local pooled row
      const int pc = lp - pr * OW;
      const long long crow0 = row0 + (long long)pr * PK;  // conv row
      const long long b = crow0 / XH;
      const int h = (int)(crow0 - b * XH);
      float a = pb;
#pragma unroll 4
      for (int i = 0; i < PK; ++i)
#pragma unroll 4
        for (int j = 0; j < PK; ++j) {
          const int lr = (pr * PK + i) * XW + pc * PK + j;
          a += pw[i * PK + j] * sc[lr * 64 + n];
        }
      // M % 64 == 0 (launcher precondition) — every tile is full
      const long long pm = (b * (XH / PK) + h / PK) * OW + pc;
      stf(pout + pm * ldC + n, sigmoidf_dev(a));
    }
  }
}

// Small-K GEMM (K <= 128, N <= 64, materialized A, Bpre given): the
// K-loop pipeline never warms at 2 iterations, so the general kernel
// pays its whole prologue per 64-row tile.  Here the B image is staged
// to LDS ONCE per workgroup and the WG grid-strides over M-tiles,
// loading each tile's A fragments STRAIGHT into MFMA registers (16B
// per lane, no LDS for A) with the next tile's loads issued before the
// current tile's MFMA block — the M-axis, not the K-axis, carries the
// pipeline.  Used by the Cin=3 stage-0 forward (K=96) and the
// LeNet-shape probe.
template <typename act_t>
__global__ __launch_bounds__(256) void k_gemm_smallk(
    const act_t* __restrict__ A, const __bf16* __restrict__ Bpre,
    const float* __restrict__ bias, act_t* __restrict__ C, long long M,
    int K, int N, int ldA, int ldC, int epilogue,
    const float* __restrict__ pw, act_t* __restrict__ pout, int PK, int XH,
    int XW) {
  __shared__ __bf16 Bs[64][136 + 8];
  __shared__ float sc[64 * 64];  // mode-3 act tile
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wv = tid >> 6;
  // stage B once: [N][K] bf16 rows, 16B chunks
  for (int idx = tid * 8; idx < N * K; idx += 256 * 8) {
    const int n = idx / K;
    const int k = idx - n * K;
    *reinterpret_cast<uint4*>(&Bs[n][k]) =
        *reinterpret_cast<const uint4*>(Bpre + (long long)n * K + k);
  }
  __syncthreads();
  const int nf = N / 16;
  const int kf = (K + 31) / 32;      // MFMA K-steps (K % 8 == 0; tail
                                     // spans read zero-padded A/B rows)
  const long long mtiles = M / 64;   // M % 64 == 0 (launcher-enforced)
  const int row = wv * 16 + (lane & 15);
  const int kq = (lane >> 4) * 8;

  bf16x8 a_cur[4], a_nxt[4];         // kf <= 4
  auto load_a = [&](long long mt, bf16x8* dst) {
    const act_t* src = A + (mt * 64 + row) * (long long)ldA + kq;
#pragma unroll
    for (int kk = 0; kk < 4; ++kk)
      if (kk < kf) {
        // in-bounds by construction: kq + kk*32 + 8 <= K <= ldA
        // (K % 32 == 0, launcher-enforced)
        if constexpr (std::is_same<act_t, bf16>::value) {
          // one 16B load straight into the fragment
          dst[kk] = *reinterpret_cast<const bf16x8*>(src + kk * 32);
        } else {
          // fp32/fp16 activations: load + convert (the general kernels
          // do the same at their LDS stage)
          float v8[8];
          ld8v(src + kk * 32, v8);
          union {
            __bf16 h[8];
            bf16x8 v;
          } t2;
#pragma unroll
          for (int u = 0; u < 8; ++u) t2.h[u] = (__bf16)v8[u];
          dst[kk] = t2.v;
        }
      }
  };

  long long t = blockIdx.x;
  if (t < mtiles) load_a(t, a_cur);
  for (; t < mtiles; t += gridDim.x) {
    const long long tn = t + gridDim.x;
    if (tn < mtiles) load_a(tn, a_nxt);
    f32x4 acc[4];
#pragma unroll
    for (int f = 0; f < 4; ++f) acc[f] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int kk = 0; kk < 4; ++kk)
      if (kk < kf) {
#pragma unroll
        for (int f = 0; f < 4; ++f)
          if (f < nf) {
            const bf16x8 bf =
                frag_from_lds(Bs[f * 16 + (lane & 15)], kk * 32 + kq);
            acc[f] =
                __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_cur[kk], bf,
                                                        acc[f], 0, 0, 0);
          }
      }
    // epilogue (same layouts as k_gemm)
    const int ccol = lane & 15;
    const int crow = wv * 16 + (lane >> 4) * 4;
    const long long m0 = t * 64;
#pragma unroll
    for (int f = 0; f < 4; ++f)
      if (f < nf) {
        const int n = f * 16 + ccol;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          float v = acc[f][r];
          if (epilogue >= 1) v = sigmoidf_dev(v + bias[n]);
          stf(C + (m0 + crow + r) * ldC + n, v);
          if (epilogue == 3) sc[(crow + r) * 64 + n] = v;
        }
      }
    if (epilogue == 3) {
      __syncthreads();
      const int OW = XW / PK;
      const long long row0 = m0 / XW;
      const int npos = 64 / (PK * PK);
      const float pb = pw[PK * PK];
      for (int o = tid; o < npos * N; o += 256) {
        const int lp = o / N;
        const int n = o - lp * N;
        const int pr = lp / OW;
        const int pc = lp - pr * OW;
        const long long crow0 = row0 + (long long)pr * PK;
        const long long b = crow0 / XH;
        const int h = (int)(crow0 - b * XH);
        float a = pb;
        for (int i = 0; i < PK; ++i)
          for (int j = 0; j < PK; ++j)
            a += pw[i * PK + j] *
                 sc[((pr * PK + i) * XW + pc * PK + j) * 64 + n];
        const long long pm = (b * (XH / PK) + h / PK) * OW + pc;
        stf(pout + pm * ldC + n, sigmoidf_dev(a));
      }
      __syncthreads();  // sc reuse on the next m-tile
    }
#pragma unroll
    for (int kk = 0; kk < 4; ++kk) a_cur[kk] = a_nxt[kk];
  }
}

// Weight-grad GEMM: dW[KcP][N] += cols^T[Kc x M-slice] @ dpre[M-slice x N].
// Both operands are transpose-staged into the [row][k=m] LDS image; the
// M dimension is the MFMA K axis.  Grid: (kc-tiles) x (n-tiles) x MS
// M-slices; fp32 hardware atomics combine slices.
// imx != null: the cols operand is the im2col view of NHWC imx
// (XC % 8 == 0, launcher-enforced): each thread's kc-span is FIXED, so
// its (i, j, ci) decode hoists out of the M-walk entirely; per chunk only
// the (b, oh, ow) row decode (two magic divisions) remains, and the
// gather is a clamped unconditional 16B load + select-to-zero.
// db != null: the conv BIAS grad db[n] += sum_m dpre[m][n] is folded in
// (replacing the separate k_colsum/k_colsum_fin passes): every dpre
// element is staged through write_lds exactly once per (kct == 0) WG,
// accumulated per-thread and LDS-reduced after the MFMA loop.
template <typename act_t>
__device__ __forceinline__ void wgrad_body(
    unsigned bidx, const act_t* __restrict__ cols,
    const act_t* __restrict__ dpre,
    float* __restrict__ dW, float* __restrict__ part, long long M, int KcP,
    int N, int MS, const act_t* __restrict__ imx, int XH, int XW, int XC,
    int XK, int XP, float* __restrict__ db, unsigned long long fd_cin,
    unsigned long long fd_k, unsigned long long fd_xw,
    unsigned long long fd_xh) {
  __shared__ GemmLds Lb[2];  // double-buffered M-chunks
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wv = tid >> 6;
  const int ntiles = (N + BN - 1) / BN;
  const int ktiles = (KcP + BM - 1) / BM;
  const int kct = bidx % max(1, ktiles);
  const int rest = bidx / max(1, ktiles);
  const int ntile = rest % ntiles;
  const int slice = rest / ntiles;
  const int kc0 = kct * BM;
  const int n0 = ntile * BN;
  const int nf = min(BN, N - n0) / 16;

  const long long m_lo = (M * slice) / MS;
  const long long m_hi = (M * (slice + 1)) / MS;

  f32x4 acc[BN / 16];
#pragma unroll
  for (int f = 0; f < BN / 16; ++f) acc[f] = {0.f, 0.f, 0.f, 0.f};

  const int row_s = tid >> 2;        // source row m (64 rows, 4 thr each)
  const int cq = (tid & 3) * 16;     // 16 columns per thread
  // T14-style pipeline (see k_gemm): prefetch the next M-chunk into
  // registers under the MFMAs, write to LDS after the read barrier.
  float rc[2][8];  // cols prefetch
  float rd[2][8];  // dpre prefetch

  const int iKc = XK * XK * XC;
  // implicit-A: this thread's kc spans are FIXED — hoist the (i, j, ci)
  // decode out of the M-walk (two magic divisions per span, once)
  int hi_i[2], hi_j[2], hi_ci[2];
  bool hi_in[2];
  if (imx != nullptr) {
#pragma unroll
    for (int h = 0; h < 2; ++h) {
      const int kc = kc0 + cq + h * 8;
      const unsigned p = fdiv((unsigned)kc, fd_cin);
      hi_ci[h] = kc - (int)p * XC;
      const unsigned pi = fdiv(p, fd_k);
      hi_j[h] = (int)(p - pi * (unsigned)XK);
      hi_i[h] = (int)pi;
      hi_in[h] = kc < iKc;
    }
  }
  // fused colsum partials: thread owns columns [n0+cq, +16)
  float dbacc[16];
  const bool do_db = db != nullptr && kct == 0;
#pragma unroll
  for (int u = 0; u < 16; ++u) dbacc[u] = 0.f;

  auto load_regs = [&](long long mt) {
    const long long m = mt + row_s;
    // m < m_hi implies m < M (m_hi = M*(slice+1)/MS <= M); clamped
    // unconditional loads + select-to-zero keep the K-loop load pipeline
    // intact (a guarded load makes hipcc branch around it).  KcP % 32 == 0
    // and N % 16 == 0, so an 8-span is either fully in or fully out.
    const bool ok = m < m_hi;
    const long long mm = ok ? m : m_hi - 1;
    int ib = 0, ioh = 0, iow = 0;
    if (imx != nullptr) {
      const unsigned bhq = fdiv((unsigned)mm, fd_xw);
      iow = (int)((unsigned)mm - bhq * (unsigned)XW);
      const unsigned bq = fdiv(bhq, fd_xh);
      ioh = (int)(bhq - bq * (unsigned)XH);
      ib = (int)bq;
    }
#pragma unroll
    for (int h = 0; h < 2; ++h) {
      const int c = cq + h * 8;
      if (imx != nullptr) {
        const int ih = ioh + hi_i[h] - XP;
        const int iw = iow + hi_j[h] - XP;
        const int ihc = min(max(ih, 0), XH - 1);
        const int iwc = min(max(iw, 0), XW - 1);
        ld8v(imx + (((long long)ib * XH + ihc) * XW + iwc) * XC + hi_ci[h],
             rc[h]);
        if (!(ok && hi_in[h] && ih == ihc && iw == iwc)) {
#pragma unroll
          for (int u = 0; u < 8; ++u) rc[h][u] = 0.f;
        }
      } else {
        const bool cok = (kc0 + c + 8) <= KcP;
        const int ccl = cok ? kc0 + c : KcP - 8;
        ld8v(cols + mm * KcP + ccl, rc[h]);
        if (!(ok && cok)) {
#pragma unroll
          for (int u = 0; u < 8; ++u) rc[h][u] = 0.f;
        }
      }
      {
        const bool nok = (n0 + c + 8) <= N;
        const int ncl = nok ? n0 + c : N - 8;
        ld8v(dpre + mm * N + ncl, rd[h]);
        if (!(ok && nok)) {
#pragma unroll
          for (int u = 0; u < 8; ++u) rd[h][u] = 0.f;
        }
      }
    }
  };

  // XOR-swizzled transpose staging: the column write (one element per
  // LDS row, all threads of a c-group on the same bank — any row pitch
  // that keeps 16B-aligned fragment reads collapses mod 32) measured
  // 22% LDSBankConflict.  Swizzling the 8-element m-group by the row
  // index spreads the writes; fragment reads stay single 16B ds_reads
  // (group-granular swizzle).
  auto sw = [](int k, int m) {
    // fold BOTH k&7 and k>>3 into the group swizzle: the simultaneous
    // writers share k mod 8 (their k differ by 16), so low bits alone
    // would leave them on one bank
    return (((m >> 3) ^ (k & 7) ^ ((k >> 3) & 7)) & 7) * 8 + (m & 7);
  };
  auto write_lds = [&](GemmLds& L) {
#pragma unroll
    for (int h = 0; h < 2; ++h) {
      const int c = cq + h * 8;
#pragma unroll
      for (int u = 0; u < 8; ++u) {
        L.As[c + u][sw(c + u, row_s)] = (__bf16)rc[h][u];
        L.Bs[c + u][sw(c + u, row_s)] = (__bf16)rd[h][u];
        if (do_db) dbacc[h * 8 + u] += rd[h][u];
      }
    }
  };

  // Double-buffered pipeline, one barrier per M-chunk (see k_gemm)
  load_regs(m_lo);
  write_lds(Lb[0]);
  if (m_lo + BK < m_hi) load_regs(m_lo + BK);
  __syncthreads();
  int p = 0;
  for (long long mt = m_lo; mt < m_hi; mt += BK, p ^= 1) {
    auto& L = Lb[p];
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      const int ar = wv * 16 + (lane & 15);
      const int m8 = kk * 32 + (lane >> 4) * 8;  // 8-aligned: sw() keeps
                                                 // the span one 16B read
      const bf16x8 a0 = frag_from_lds(L.As[ar], sw(ar, m8));
#pragma unroll
      for (int f = 0; f < BN / 16; ++f) {
        if (f < nf) {
          const int br = f * 16 + (lane & 15);
          const bf16x8 bf = frag_from_lds(L.Bs[br], sw(br, m8));
          acc[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, bf, acc[f], 0,
                                                           0, 0);
        }
      }
    }
    if (mt + BK < m_hi) {
      write_lds(Lb[p ^ 1]);
      if (mt + 2 * BK < m_hi) load_regs(mt + 2 * BK);
    }
    __syncthreads();
  }

  // fused colsum epilogue: fold the per-thread dpre partials across the
  // 64 staging rows (LDS reuse of the now-idle A tile) and combine into
  // db with one atomic per column per (kct==0, slice) WG.
  if (do_db) {
    float* sc = reinterpret_cast<float*>(&Lb[0]);  // [64 rows][64 cols] fp32
#pragma unroll
    for (int u = 0; u < 16; ++u) sc[row_s * 64 + cq + u] = dbacc[u];
    __syncthreads();
    if (tid < 64 && (n0 + tid) < N) {
      float s = 0.f;
      for (int r = 0; r < 64; ++r) s += sc[r * 64 + tid];
      if (s != 0.f) unsafeAtomicAdd(&db[n0 + tid], s);
    }
    __syncthreads();  // scratch reads done before any later reuse
  }

  const int crow = wv * 16 + (lane >> 4) * 4;  // kc within tile
  const int ccol = lane & 15;                  // n within fragment
  // part != nullptr: slab mode — every (kct, ntile) pair owns a disjoint
  // region of slab `slice`, so these are plain stores and the slice count
  // is decoupled from any atomic-combine cost (k_wsum folds the slabs).
  float* const slab =
      part ? part + (long long)slice * KcP * N : nullptr;
#pragma unroll
  for (int f = 0; f < BN / 16; ++f) {
    if (f < nf) {
      const int n = n0 + f * 16 + ccol;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int kc = kc0 + crow + r;
        if (kc < KcP && n < N) {
          const float v = acc[f][r];
          if (slab)
            slab[(long long)kc * N + n] = v;
          else if (v != 0.f)
            unsafeAtomicAdd(&dW[(long long)kc * N + n], v);
        }
      }
    }
  }
}

template <typename act_t>
__global__ __launch_bounds__(256) void k_wgrad_gemm(
    const act_t* __restrict__ cols, const act_t* __restrict__ dpre,
    float* __restrict__ dW, float* __restrict__ part, long long M, int KcP,
    int N, int MS, const act_t* __restrict__ imx, int XH, int XW, int XC,
    int XK, int XP, float* __restrict__ db, unsigned long long fd_cin,
    unsigned long long fd_k, unsigned long long fd_xw,
    unsigned long long fd_xh) {
  wgrad_body<act_t>(blockIdx.x, cols, dpre, dW, part, M, KcP, N, MS, imx,
                    XH, XW, XC, XK, XP, db, fd_cin, fd_k, fd_xw, fd_xh);
}

// All conv stages' weight-grad GEMMs in ONE launch: three sequential
// ~500-WG launches each run at ~2 WGs/CU with the per-iteration stall
// exposed; the combined grid fills the chip and lets the stages'
// latencies hide each other.  Stage found by cumulative block offsets.
struct WgDesc {
  int n;                       // stages (<= 8)
  long long blk_cum[9];        // cumulative grid blocks
  const void* a[8];            // cols (materialized) or imx (implicit)
  const void* dpre[8];
  float* dW[8];
  float* db[8];
  long long M[8];
  int KcP[8], N[8], MS[8], implicit[8];
  int XH[8], XW[8], XC[8], XK[8], XP[8];
  unsigned long long fd_cin[8], fd_k[8], fd_xw[8], fd_xh[8];
};
template <typename act_t>
__global__ __launch_bounds__(256) void k_wgrad_multi(WgDesc d) {
  int s = 0;
  while (blockIdx.x >= d.blk_cum[s + 1]) ++s;
  const unsigned bidx = (unsigned)(blockIdx.x - d.blk_cum[s]);
  wgrad_body<act_t>(
      bidx, (const act_t*)d.a[s], (const act_t*)d.dpre[s], d.dW[s],
      nullptr, d.M[s], d.KcP[s], d.N[s], d.MS[s],
      d.implicit[s] ? (const act_t*)d.a[s] : nullptr, d.XH[s], d.XW[s],
      d.XC[s], d.XK[s], d.XP[s], d.db[s], d.fd_cin[s], d.fd_k[s],
      d.fd_xw[s], d.fd_xh[s]);
}

// Fold MS weight-grad slabs into dW (+=): 8 fp32 per thread, the slab
// axis walked with 2 streams in flight.
__global__ __launch_bounds__(256) void k_wsum(const float* __restrict__ part,
                                              float* __restrict__ dW,
                                              long long total, int MS) {
  const long long i8 = ((long long)blockIdx.x * 256 + threadIdx.x) * 8;
  if (i8 >= total) return;
  float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  int sl = 0;
  for (; sl + 2 <= MS; sl += 2) {
    float a[8], b[8];
    ld8v(part + sl * total + i8, a);
    ld8v(part + (sl + 1) * total + i8, b);
#pragma unroll
    for (int u = 0; u < 8; ++u) acc[u] += a[u] + b[u];
  }
  if (sl < MS) {
    float a[8];
    ld8v(part + sl * total + i8, a);
#pragma unroll
    for (int u = 0; u < 8; ++u) acc[u] += a[u];
  }
#pragma unroll
  for (int u = 0; u < 8; ++u) dW[i8 + u] += acc[u];
}

// Column-sum for the conv bias grad: db[n] += sum_m dpre[m][n].
// Each thread owns 8 consecutive elements of the flat [M*N] stream (one
// 16B load per iteration); since the grid stride is a multiple of N, the
// 8 columns a thread sees are FIXED, so it keeps 8 private partials and
// the block tree-reduces per column at the end.
template <typename act_t>
__global__ __launch_bounds__(256) void k_colsum(const act_t* __restrict__ dpre,
                                                float* __restrict__ part,
                                                long long M, int N, int G) {
  const int tid = threadIdx.x;
  const long long total = M * N;
  const long long stride = (long long)G * 256 * 8;  // multiple of N (N%8==0,
                                                    // 2048%N==0 for N<=256)
  long long flat = ((long long)blockIdx.x * 256 + tid) * 8;
  float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  // 4 independent streams in flight: a single-stream loop is a load
  // latency chain at the grid sizes this runs at (~1 workgroup/CU).
  for (; flat + 3 * stride + 8 <= total; flat += 4 * stride) {
    float v[4][8];
#pragma unroll
    for (int q = 0; q < 4; ++q) ld8v(dpre + flat + q * stride, v[q]);
#pragma unroll
    for (int q = 0; q < 4; ++q)
#pragma unroll
      for (int u = 0; u < 8; ++u) acc[u] += v[q][u];
  }
  for (; flat + 8 <= total; flat += stride) {
    float v8[8];
    ld8v(dpre + flat, v8);
#pragma unroll
    for (int u = 0; u < 8; ++u) acc[u] += v8[u];
  }
  __shared__ float sred[256][8];
#pragma unroll
  for (int u = 0; u < 8; ++u) sred[tid][u] = acc[u];
  __syncthreads();
  // threads with the same n0 sit N/8 apart; fold the thread axis
  for (int off = 128; off >= N / 8; off >>= 1) {
    if (tid < off) {
#pragma unroll
      for (int u = 0; u < 8; ++u) sred[tid][u] += sred[tid + off][u];
    }
    __syncthreads();
  }
  // per-workgroup partials to scratch (atomics to the handful of db
  // cache lines serialize globally at ~3ns/op — measured to dominate
  // this kernel at any G; k_colsum_fin combines the partials instead)
  if (tid < N / 8) {
    float* dst = part + (long long)blockIdx.x * N + tid * 8;
#pragma unroll
    for (int u = 0; u < 8; ++u) dst[u] = sred[tid][u];
  }
}

// Combine the [G][N] column partials into db (+=).  One workgroup: 256/N
// lanes per column sum a G/L stride each (4 streams in flight), then a
// small LDS fold.  N > 256 falls back to one thread per column.
__global__ __launch_bounds__(256) void k_colsum_fin(
    const float* __restrict__ part, float* __restrict__ db, int N, int G) {
  const int tid = threadIdx.x;
  if (N >= 256) {
    for (int n = tid; n < N; n += 256) {
      float t = 0.f;
      for (int g = 0; g < G; ++g) t += part[(long long)g * N + n];
      db[n] += t;
    }
    return;
  }
  const int L = 256 / N;  // lanes per column (N % 8 == 0)
  const int n = tid % N;
  const int sl = tid / N;
  float s = 0.f;
  if (sl < L) {
    int g = sl;
    for (; g + 3 * L < G; g += 4 * L)
      s += part[(long long)g * N + n] + part[(long long)(g + L) * N + n] +
           part[(long long)(g + 2 * L) * N + n] +
           part[(long long)(g + 3 * L) * N + n];
    for (; g < G; g += L) s += part[(long long)g * N + n];
  }
  __shared__ float red[256];
  red[tid] = s;
  __syncthreads();
  if (tid < N) {
    float tot = 0.f;
    for (int s2 = 0; s2 < L; ++s2) tot += red[tid + s2 * N];
    db[tid] += tot;
  }
}

// ---------------------------------------------------------------------------
// col2im (gather form) fused with the pool-output sigmoid backward of the
// PREVIOUS block:  for each input pixel of this conv,
//   g = sum_{i,j valid} dcols[(b, h+P-i, w+P-j)][(i*K+j)*Cin+ci]
//   out[b,h,w,ci] = g * pout*(1-pout)        (pout = that pixel's value)
// When pout == nullptr the sigmoid factor is skipped (plain dX).
// ---------------------------------------------------------------------------
template <typename act_t, bool FAST>
__global__ void k_col2im_sigbwd(const act_t* __restrict__ dcols,
                                const act_t* __restrict__ pout,
                                act_t* __restrict__ out, int B, int H, int W,
                                int Cin, int K, int P, int KcP,
                                unsigned long long fd_cpr,
                                unsigned long long fd_w,
                                unsigned long long fd_h) {
  // 8 consecutive ci per thread (Cin % 8 == 0 for every caller): the K*K
  // gather loads become 8-wide vector loads.
  const long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long total = (long long)B * H * W * Cin / 8;
  if (idx >= total) return;
  int ci0, w, h, b;
  if (FAST) {
    const unsigned n = (unsigned)idx;
    const unsigned cpr = (unsigned)(Cin / 8);
    const unsigned tq = fdiv(n, fd_cpr);
    ci0 = (int)(n - tq * cpr) * 8;
    const unsigned hq = fdiv(tq, fd_w);
    w = (int)(tq - hq * (unsigned)W);
    const unsigned bq = fdiv(hq, fd_h);
    h = (int)(hq - bq * (unsigned)H);
    b = (int)bq;
  } else {
    ci0 = (int)((idx * 8) % Cin);
    long long t = (idx * 8) / Cin;
    w = (int)(t % W);
    t /= W;
    h = (int)(t % H);
    b = (int)(t / H);
  }
  float g[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  for (int i = 0; i < K; ++i) {
    const int oh = h + P - i;
    if (oh < 0 || oh >= H) continue;
    for (int j = 0; j < K; ++j) {
      const int ow = w + P - j;
      if (ow < 0 || ow >= W) continue;
      const long long m = ((long long)b * H + oh) * W + ow;
      float v8[8];
      ld8v(dcols + m * KcP + (i * K + j) * Cin + ci0, v8);
#pragma unroll
      for (int u = 0; u < 8; ++u) g[u] += v8[u];
    }
  }
  act_t o8[8];
  if (pout != nullptr) {
    float p8[8];
    ld8v(pout + idx * 8, p8);
#pragma unroll
    for (int u = 0; u < 8; ++u)
      o8[u] = (act_t)(g[u] * p8[u] * (1.0f - p8[u]));
  } else {
#pragma unroll
    for (int u = 0; u < 8; ++u) o8[u] = (act_t)g[u];
  }
  if (sizeof(act_t) == 2)
    *reinterpret_cast<uint4*>(out + idx * 8) =
        *reinterpret_cast<const uint4*>(o8);
  else
#pragma unroll
    for (int u = 0; u < 8; ++u) out[idx * 8 + u] = o8[u];
}

// ---------------------------------------------------------------------------
// Trainable pool (NHWC, shared KxK kernel, stride == K, scalar bias; the
// framework's generalization of the reference's trainable 4x4 pool).
// ---------------------------------------------------------------------------
// 8 consecutive channels per thread (C % 8 == 0): all loads/stores 16B.
template <typename act_t>
__global__ void k_pool_fwd(const act_t* __restrict__ a,
                           const float* __restrict__ pw,  // [K*K] then bias
                           act_t* __restrict__ pout, int B, int H, int W,
                           int C, int K) {
  const int OH = H / K, OW = W / K;
  const long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long total = (long long)B * OH * OW * C / 8;
  if (idx >= total) return;
  const int c0 = (int)((idx * 8) % C);
  long long t = (idx * 8) / C;
  const int q = (int)(t % OW);
  t /= OW;
  const int p = (int)(t % OH);
  const int b = (int)(t / OH);
  float acc[8];
#pragma unroll
  for (int u = 0; u < 8; ++u) acc[u] = pw[K * K];
  for (int i = 0; i < K; ++i)
    for (int j = 0; j < K; ++j) {
      float av[8];
      ld8v(a + (((long long)b * H + p * K + i) * W + q * K + j) * C + c0, av);
      const float wv = pw[i * K + j];
#pragma unroll
      for (int u = 0; u < 8; ++u) acc[u] += wv * av[u];
    }
  act_t out[8];
#pragma unroll
  for (int u = 0; u < 8; ++u) out[u] = (act_t)sigmoidf_dev(acc[u]);
  if (sizeof(act_t) == 2)
    *reinterpret_cast<uint4*>(pout + idx * 8) =
        *reinterpret_cast<const uint4*>(out);
  else
#pragma unroll
    for (int u = 0; u < 8; ++u) pout[idx * 8 + u] = out[u];
}

// dpre_conv[b,h,w,c] = dppre[b,h/K,w/K,c] * pw[h%K,w%K] * a*(1-a)
template <typename act_t>
__global__ void k_pool_bwd(const act_t* __restrict__ dppre,
                           const act_t* __restrict__ a,
                           const float* __restrict__ pw,
                           act_t* __restrict__ dapre, int B, int H, int W,
                           int C, int K) {
  const int OH = H / K, OW = W / K;
  const long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long total = (long long)B * H * W * C / 8;
  if (idx >= total) return;
  const int c0 = (int)((idx * 8) % C);
  long long t = (idx * 8) / C;
  const int w = (int)(t % W);
  t /= W;
  const int h = (int)(t % H);
  const int b = (int)(t / H);
  float dv[8], av[8];
  ld8v(dppre + (((long long)b * OH + h / K) * OW + w / K) * C + c0, dv);
  ld8v(a + idx * 8, av);
  const float wv = pw[(h % K) * K + (w % K)];
  act_t out[8];
#pragma unroll
  for (int u = 0; u < 8; ++u)
    out[u] = (act_t)(dv[u] * wv * av[u] * (1.0f - av[u]));
  if (sizeof(act_t) == 2)
    *reinterpret_cast<uint4*>(dapre + idx * 8) =
        *reinterpret_cast<const uint4*>(out);
  else
#pragma unroll
    for (int u = 0; u < 8; ++u) dapre[idx * 8 + u] = out[u];
}

// pool wgrad: dpw[i,j] += sum dppre[b,p,q,c] * a[b,pK+i,qK+j,c];
// bias += sum dppre.  Grid-stride, per-thread regs, wave reduce, atomics.
template <typename act_t>
__global__ __launch_bounds__(256) void k_pool_wgrad(
    const act_t* __restrict__ dppre, const act_t* __restrict__ a,
    float* __restrict__ dpw, int B, int H, int W, int C, int K, int G) {
  // Two items in flight per iteration: the second item's loads issue
  // before the first item's FMAs (the K*K+1 loads per item are otherwise
  // a serial latency chain).
  const int OH = H / K, OW = W / K;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  float acc[4 * 4];  // K <= 4 supported
  for (int w = 0; w < K * K; ++w) acc[w] = 0.f;
  float bacc = 0.f;
  const long long N = (long long)B * OH * OW * C;
  const long long stride = (long long)G * 256;

  auto decode_load = [&](long long it, float* d, float* av) {
    const int c = (int)(it % C);
    long long t = it / C;
    const int q = (int)(t % OW);
    t /= OW;
    const int p = (int)(t % OH);
    const int b = (int)(t / OH);
    *d = ldf(dppre + it);
    for (int i = 0; i < K; ++i)
      for (int j = 0; j < K; ++j)
        av[i * K + j] =
            ldf(a + (((long long)b * H + p * K + i) * W + q * K + j) * C + c);
  };

  for (long long it = (long long)blockIdx.x * 256 + tid; it < N;
       it += 2 * stride) {
    float d1, av1[16], d2 = 0.f, av2[16];
    decode_load(it, &d1, av1);
    const long long it2 = it + stride;
    if (it2 < N) decode_load(it2, &d2, av2);
    bacc += d1;
    for (int w = 0; w < K * K; ++w) acc[w] += d1 * av1[w];
    if (it2 < N) {
      bacc += d2;
      for (int w = 0; w < K * K; ++w) acc[w] += d2 * av2[w];
    }
  }
  __shared__ float red[4][17];
  const int wv = tid >> 6;
  for (int w = 0; w < K * K; ++w) {
    float v = acc[w];
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
    if (lane == 0) red[wv][w] = v;
  }
  {
    float v = bacc;
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
    if (lane == 0) red[wv][16] = v;
  }
  __syncthreads();
  if (tid < K * K)
    unsafeAtomicAdd(&dpw[tid],
                    red[0][tid] + red[1][tid] + red[2][tid] + red[3][tid]);
  if (tid == K * K)
    unsafeAtomicAdd(&dpw[K * K],
                    red[0][16] + red[1][16] + red[2][16] + red[3][16]);
}

// Vectorized pool wgrad for C % 8 == 0 (every DeepCNN config): each
// thread owns 8 consecutive channels, so all K*K+1 loads per item are
// 16B ld8v instead of scalar 2B loads (scalar bf16 access is
// address-issue-bound on CDNA4 — this is the k_pool_fwd idiom applied
// to the reduction).  acc[w] folds the 8 channels immediately (the pool
// weight is shared across channels).  Wave + LDS pre-reduce, 17 atomics
// per workgroup, as in the scalar kernel above.
template <typename act_t, int KK>
__global__ __launch_bounds__(256) void k_pool_wgrad8(
    const act_t* __restrict__ dppre, const act_t* __restrict__ a,
    float* __restrict__ dpw, int B, int H, int W, int C, int G) {
  const int OH = H / KK, OW = W / KK;
  const int C8 = C / 8;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  float acc[KK * KK];
#pragma unroll
  for (int w = 0; w < KK * KK; ++w) acc[w] = 0.f;
  float bacc = 0.f;
  const long long N8 = (long long)B * OH * OW * C8;
  const long long stride = (long long)G * 256;

  auto decode_load = [&](long long it, float* d, float (*av)[8]) {
    const int c0 = (int)(it % C8) * 8;
    long long t = it / C8;
    const int q = (int)(t % OW);
    t /= OW;
    const int p = (int)(t % OH);
    const int b = (int)(t / OH);
    ld8v(dppre + (((long long)b * OH + p) * OW + q) * C + c0, d);
#pragma unroll
    for (int i = 0; i < KK; ++i)
#pragma unroll
      for (int j = 0; j < KK; ++j)
        ld8v(a + (((long long)b * H + p * KK + i) * W + q * KK + j) * C + c0,
             av[i * KK + j]);
  };

  for (long long it = (long long)blockIdx.x * 256 + tid; it < N8;
       it += 2 * stride) {
    float d1[8], av1[KK * KK][8], d2[8], av2[KK * KK][8];
    decode_load(it, d1, av1);
    const long long it2 = it + stride;
    if (it2 < N8) decode_load(it2, d2, av2);
#pragma unroll
    for (int u = 0; u < 8; ++u) bacc += d1[u];
#pragma unroll
    for (int w = 0; w < KK * KK; ++w)
#pragma unroll
      for (int u = 0; u < 8; ++u) acc[w] += d1[u] * av1[w][u];
    if (it2 < N8) {
#pragma unroll
      for (int u = 0; u < 8; ++u) bacc += d2[u];
#pragma unroll
      for (int w = 0; w < KK * KK; ++w)
#pragma unroll
        for (int u = 0; u < 8; ++u) acc[w] += d2[u] * av2[w][u];
    }
  }
  __shared__ float red[4][KK * KK + 1];
  const int wv = tid >> 6;
#pragma unroll
  for (int w = 0; w < KK * KK; ++w) {
    float v = acc[w];
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
    if (lane == 0) red[wv][w] = v;
  }
  {
    float v = bacc;
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
    if (lane == 0) red[wv][KK * KK] = v;
  }
  __syncthreads();
  if (tid <= KK * KK)
    unsafeAtomicAdd(&dpw[tid],
                    red[0][tid] + red[1][tid] + red[2][tid] + red[3][tid]);
}

// Fused pool wgrad + backward (C % 8 == 0): the two kernels read the
// same dppre and acts; one pass computes the K*K+1 weight-grad partials
// AND writes the expanded conv preact grad dapre = d * pw[ij] * a *
// (1-a) IN PLACE over the activation (same-thread read-then-write per
// element, so the overwrite is safe).  Halves the pool backward's HBM
// reads and removes a launch.
template <typename act_t, int KK>
__global__ __launch_bounds__(256) void k_pool_wbwd8(
    const act_t* __restrict__ dppre, const act_t* __restrict__ a,
    const float* __restrict__ pw, act_t* __restrict__ dapre,
    float* __restrict__ dpw, int B, int H, int W, int C, int G) {
  const int OH = H / KK, OW = W / KK;
  const int C8 = C / 8;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  float acc[KK * KK];
#pragma unroll
  for (int w = 0; w < KK * KK; ++w) acc[w] = 0.f;
  float bacc = 0.f;
  const long long N8 = (long long)B * OH * OW * C8;
  const long long stride = (long long)G * 256;

  for (long long it = (long long)blockIdx.x * 256 + tid; it < N8;
       it += stride) {
    const int c0 = (int)(it % C8) * 8;
    long long t = it / C8;
    const int q = (int)(t % OW);
    t /= OW;
    const int p = (int)(t % OH);
    const int b = (int)(t / OH);
    float d[8];
    ld8v(dppre + (((long long)b * OH + p) * OW + q) * C + c0, d);
#pragma unroll
    for (int u = 0; u < 8; ++u) bacc += d[u];
#pragma unroll
    for (int i = 0; i < KK; ++i)
#pragma unroll
      for (int j = 0; j < KK; ++j) {
        const long long off =
            (((long long)b * H + p * KK + i) * W + q * KK + j) * C + c0;
        float av[8];
        ld8v(a + off, av);
        const float wv = pw[i * KK + j];
        act_t o8[8];
#pragma unroll
        for (int u = 0; u < 8; ++u) {
          acc[i * KK + j] += d[u] * av[u];
          o8[u] = (act_t)(d[u] * wv * av[u] * (1.0f - av[u]));
        }
        if (sizeof(act_t) == 2)
          *reinterpret_cast<uint4*>(dapre + off) =
              *reinterpret_cast<const uint4*>(o8);
        else
#pragma unroll
          for (int u = 0; u < 8; ++u) dapre[off + u] = o8[u];
      }
  }
  __shared__ float red[4][KK * KK + 1];
  const int wv = tid >> 6;
#pragma unroll
  for (int w = 0; w < KK * KK; ++w) {
    float v = acc[w];
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
    if (lane == 0) red[wv][w] = v;
  }
  {
    float v = bacc;
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
    if (lane == 0) red[wv][KK * KK] = v;
  }
  __syncthreads();
  if (tid <= KK * KK)
    unsafeAtomicAdd(&dpw[tid],
                    red[0][tid] + red[1][tid] + red[2][tid] + red[3][tid]);
}

// ---------------------------------------------------------------------------
// FC head (FCIN -> 10) + residual loss, general fan-in.  One block per
// sample; 16 lanes per class with shuffle reduction (the LeNet pattern).
// mode: 0 train (emit dz + loss), 1 eval (argmax + correct), 2 infer.
// ---------------------------------------------------------------------------
// dflat != null (train mode): the fc backward-data is fused in — after
// the block's dz is in LDS, the same block computes dflat[b][m] =
// (sum_k fw[k][m] dz[k]) * flat*(1-flat) for its sample (kills the
// separate k_fc_bwd launch; fw re-read comes from L2).
template <typename act_t>
__global__ __launch_bounds__(256) void k_fc_fwd(
    const act_t* __restrict__ flat, const float* __restrict__ fw,
    const float* __restrict__ fb, const int* __restrict__ labels,
    float* __restrict__ yg, float* __restrict__ dzg,
    float* __restrict__ loss_accum, int* __restrict__ correct_accum, int B,
    int FCIN, int NCLS, int mode, act_t* __restrict__ dflat) {
  __shared__ float ys[32];
  __shared__ float sq[32];
  __shared__ float dzs[32];
  const int b = blockIdx.x;
  if (b >= B) return;
  const int tid = threadIdx.x;
  const act_t* xb = flat + (long long)b * FCIN;
  if (tid < NCLS * 16) {
    const int k = tid >> 4;
    const int l = tid & 15;
    const float* wk = fw + (long long)k * FCIN;
    float p = 0.f;
    if ((FCIN % 128) == 0) {
      // 8-wide vector loads per lane: FCIN=1024 is 8 iterations of 16B
      // loads instead of 64 dependent scalar loads (the serial chain was
      // the whole kernel's latency at B blocks ~ 1/CU occupancy)
      for (int m0 = l * 8; m0 < FCIN; m0 += 128) {
        float xv[8], wv8[8];
        ld8v(xb + m0, xv);
        ld8v(wk + m0, wv8);
#pragma unroll
        for (int u = 0; u < 8; ++u) p += wv8[u] * xv[u];
      }
    } else {
      for (int m = l; m < FCIN; m += 16) p += wk[m] * ldf(xb + m);
    }
#pragma unroll
    for (int off = 8; off > 0; off >>= 1) p += __shfl_down(p, off, 16);
    if (l == 0) {
      const float v = sigmoidf_dev(p + fb[k]);
      ys[k] = v;
      if (yg != nullptr) yg[(long long)b * NCLS + k] = v;
      if (mode == 0) {
        const float d = (k == labels[b] ? 1.0f : 0.0f) - v;
        sq[k] = d * d;
        dzs[k] = d;
        dzg[(long long)b * NCLS + k] = d;
      }
    }
  }
  __syncthreads();
  if (mode == 1 && tid == 0) {
    int best = 0;
    for (int k = 1; k < NCLS; ++k)
      if (ys[k] > ys[best]) best = k;
    if (best == labels[b]) atomicAdd(correct_accum, 1);
  }
  if (mode == 0 && tid == 0 && loss_accum != nullptr) {
    float s = 0.f;
    for (int k = 0; k < NCLS; ++k) s += sq[k];
    unsafeAtomicAdd(loss_accum, sqrtf(s));
  }
  if (mode == 0 && dflat != nullptr) {
    for (int m = tid; m < FCIN; m += 256) {
      float da = 0.f;
      for (int k = 0; k < NCLS; ++k)
        da += fw[(long long)k * FCIN + m] * dzs[k];
      const float v = ldf(xb + m);
      stf(dflat + (long long)b * FCIN + m, da * v * (1.0f - v));
    }
  }
}

// dflat[b,m] = (sum_k fw[k,m] dz[b,k]) * flat*(1-flat)   (act_t out)
template <typename act_t>
__global__ void k_fc_bwd(const float* __restrict__ dzg,
                         const act_t* __restrict__ flat,
                         const float* __restrict__ fw,
                         act_t* __restrict__ dflat, int B, int FCIN,
                         int NCLS) {
  const long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= (long long)B * FCIN) return;
  const int m = (int)(idx % FCIN);
  const int b = (int)(idx / FCIN);
  float da = 0.f;
  for (int k = 0; k < NCLS; ++k)
    da += fw[(long long)k * FCIN + m] * dzg[(long long)b * NCLS + k];
  const float v = ldf(flat + idx);
  stf(dflat + idx, da * v * (1.0f - v));
}

// fc wgrad: owner thread per (k, m) pair, batch-sliced like the LeNet fc.
template <typename act_t>
__global__ __launch_bounds__(256) void k_fc_wgrad(
    const float* __restrict__ dzg, const act_t* __restrict__ flat,
    float* __restrict__ gfw, float* __restrict__ gfb, int B, int FCIN,
    int NCLS, int FS) {
  const int nw = NCLS * FCIN;
  const int blocks_per_slice = (nw + 255 + NCLS) / 256 + 1;
  const int slice = blockIdx.x / blocks_per_slice;
  const int q = (blockIdx.x % blocks_per_slice) * 256 + threadIdx.x;
  const int b_lo = (int)(((long long)B * slice) / FS);
  const int b_hi = (int)(((long long)B * (slice + 1)) / FS);
  if (q < nw) {
    const int k = q / FCIN;
    const int m = q - k * FCIN;
    float acc = 0.f;
    for (int b = b_lo; b < b_hi; ++b)
      acc += dzg[(long long)b * NCLS + k] * ldf(flat + (long long)b * FCIN + m);
    if (FS == 1)
      gfw[q] += acc;
    else
      unsafeAtomicAdd(&gfw[q], acc);
  } else if (q < nw + NCLS) {
    const int k = q - nw;
    float acc = 0.f;
    for (int b = b_lo; b < b_hi; ++b) acc += dzg[(long long)b * NCLS + k];
    if (FS == 1)
      gfb[k] += acc;
    else
      unsafeAtomicAdd(&gfb[k], acc);
  }
}

// Weight pre-cast: fp32 W[R][C] -> bf16 copy [R][C] and bf16 transpose
// [C][R], refreshed once per step so GEMM B-staging is plain 16B bf16 row
// copies instead of per-tile fp32 gather+convert(+transpose).
__global__ void k_cast_wt(const float* __restrict__ W,
                          __bf16* __restrict__ out,
                          __bf16* __restrict__ outT, int R, int C) {
  const long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= (long long)R * C) return;
  const int r = (int)(i / C);
  const int c = (int)(i - (long long)r * C);
  const __bf16 v = (__bf16)W[i];
  out[i] = v;
  outT[(long long)c * R + r] = v;
}

// Combine split-K slabs (c32 [KS][M][N] fp32) and apply the GEMM
// epilogue.  mode 0: plain store; 1: C = sigmoid(sum + bias); 2: C =
// sum * epi * (1-epi); 3: mode 1 PLUS the trainable-pool forward (one
// thread per pooled position x 8 channels — it writes the PK*PK acts it
// combined and the pooled output; no tile-alignment requirement, the
// pooling reads c32 global directly).
template <typename act_t>
__global__ void k_split_epi(const float* __restrict__ c32, int KS,
                            act_t* __restrict__ C, long long M, int N,
                            const float* __restrict__ bias,
                            const act_t* __restrict__ epi, int epilogue,
                            const float* __restrict__ pw,
                            act_t* __restrict__ pout, int PK, int XH,
                            int XW) {
  const long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long MN = M * N;
  // 2-in-flight slab reads: two independent 16B loads per iteration so
  // the KS-deep sum is not one serial latency chain
  auto slab_sum = [&](long long off, float* v8) {
#pragma unroll
    for (int u = 0; u < 8; ++u) v8[u] = 0.f;
    int s = 0;
    for (; s + 2 <= KS; s += 2) {
      float a8[8], b8[8];
      ld8v(c32 + (long long)s * MN + off, a8);
      ld8v(c32 + (long long)(s + 1) * MN + off, b8);
#pragma unroll
      for (int u = 0; u < 8; ++u) v8[u] += a8[u] + b8[u];
    }
    if (s < KS) {
      float a8[8];
      ld8v(c32 + (long long)s * MN + off, a8);
#pragma unroll
      for (int u = 0; u < 8; ++u) v8[u] += a8[u];
    }
  };
  if (epilogue == 3) {
    // one thread per CONV element x 8 channels (PK*PK more threads than
    // pooled positions — a 32-WG combine at small M measured 19 us);
    // window-aligned thread order so each pool window lives in one
    // block, acts exchanged through LDS.
    const int OW = XW / PK;
    const int n8 = N / 8;
    const int wsz = PK * PK * n8;      // threads per pool window
    (void)wsz;
    __shared__ float lds_a[256 * 8];
    // NO early return before the barrier (divergent-barrier UB in the
    // ragged last block) — inactive threads just skip the work
    const bool active = idx * 8 < MN;
    // idx = (pp * PK*PK + ij) * n8 + c8
    const int c8 = (int)(idx % n8);
    const long long t1 = idx / n8;
    const int ij = (int)(t1 % (PK * PK));
    const long long pp = t1 / (PK * PK);
    const int i = ij / PK, j = ij % PK;
    const int q = (int)(pp % OW);
    long long t = pp / OW;
    const int p = (int)(t % (XH / PK));
    const long long b = t / (XH / PK);
    const int c0 = c8 * 8;
    const long long m = ((b * XH + p * PK + i) * XW + q * PK + j);
    if (active) {
      float v8[8];
      slab_sum(m * N + c0, v8);
      act_t o8[8];
#pragma unroll
      for (int u = 0; u < 8; ++u) {
        const float a = sigmoidf_dev(v8[u] + bias[c0 + u]);
        o8[u] = (act_t)a;
        lds_a[threadIdx.x * 8 + u] = a;
      }
      if (sizeof(act_t) == 2)
        *reinterpret_cast<uint4*>(C + m * N + c0) =
            *reinterpret_cast<const uint4*>(o8);
      else
#pragma unroll
        for (int u = 0; u < 8; ++u) C[m * N + c0 + u] = o8[u];
    }
    __syncthreads();
    if (active && ij == 0) {
      // window base thread pools its PK*PK acts from LDS
      float pacc[8];
      const float pb = pw[PK * PK];
#pragma unroll
      for (int u = 0; u < 8; ++u) pacc[u] = pb;
      const int base = (int)(threadIdx.x);  // = (local pp) * wsz + c8
      for (int w2 = 0; w2 < PK * PK; ++w2) {
        const float wv = pw[w2];
#pragma unroll
        for (int u = 0; u < 8; ++u)
          pacc[u] += wv * lds_a[(base + w2 * n8) * 8 + u];
      }
      act_t po[8];
#pragma unroll
      for (int u = 0; u < 8; ++u) po[u] = (act_t)sigmoidf_dev(pacc[u]);
      if (sizeof(act_t) == 2)
        *reinterpret_cast<uint4*>(pout + pp * N + c0) =
            *reinterpret_cast<const uint4*>(po);
      else
#pragma unroll
        for (int u = 0; u < 8; ++u) pout[pp * N + c0 + u] = po[u];
    }
    return;
  }
  // modes 0/1/2: one thread per 8 consecutive elements of the [M][N]
  // stream (N % 8 == 0)
  if (idx * 8 >= MN) return;
  const long long e0 = idx * 8;
  const int n = (int)(e0 % N);
  float v8[8];
  slab_sum(e0, v8);
  act_t o8[8];
  if (epilogue == 1) {
#pragma unroll
    for (int u = 0; u < 8; ++u)
      o8[u] = (act_t)sigmoidf_dev(v8[u] + bias[n + u]);
  } else if (epilogue == 2) {
    float e8[8];
    ld8v(epi + e0, e8);
#pragma unroll
    for (int u = 0; u < 8; ++u)
      o8[u] = (act_t)(v8[u] * e8[u] * (1.0f - e8[u]));
  } else {
#pragma unroll
    for (int u = 0; u < 8; ++u) o8[u] = (act_t)v8[u];
  }
  if (sizeof(act_t) == 2)
    *reinterpret_cast<uint4*>(C + e0) = *reinterpret_cast<const uint4*>(o8);
  else
#pragma unroll
    for (int u = 0; u < 8; ++u) C[e0 + u] = o8[u];
}

// Channel-pad: x[B*HW][Cin] -> x8[B*HW][8] with zeros in channels >= Cin
// (Cin < 8).  Lets a Cin=3 input stage run the implicit-im2col GEMM fast
// path (which needs Cin % 8 == 0) instead of materializing a 50 MB cols
// buffer: 8-channel padded input is ~4 MB at bs=256 and L2-resident for
// every re-gather.  One thread per pixel, one 16B store.
template <typename act_t>
__global__ void k_pad_channels(const act_t* __restrict__ x,
                               act_t* __restrict__ x8, long long npix,
                               int Cin) {
  const long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= npix) return;
  act_t out[8];
#pragma unroll
  for (int u = 0; u < 8; ++u)
    out[u] = u < Cin ? x[i * Cin + u] : (act_t)0.f;
  if (sizeof(act_t) == 2)
    *reinterpret_cast<uint4*>(x8 + i * 8) =
        *reinterpret_cast<const uint4*>(out);
  else
#pragma unroll
    for (int u = 0; u < 8; ++u) x8[i * 8 + u] = out[u];
}

// Remap the 8-padded weight-grad image dW8[(p*8+ci)][Cout] back into the
// model's [(p*Cin+ci)][Cout] flat-gradient layout (+=), zeroing dW8 for
// the next step.  Pad-channel rows of dW8 are never touched (their cols
// values are identically zero and the wgrad atomic skips zeros).
__global__ void k_remap_dw8(float* __restrict__ dW8,
                            float* __restrict__ dW, int KK, int Cin,
                            int Cout) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= KK * Cin * Cout) return;
  const int c = i % Cout;
  const int t = i / Cout;
  const int ci = t % Cin;
  const int p = t / Cin;
  float* src = dW8 + ((long long)(p * 8 + ci) * Cout + c);
  dW[i] += *src;
  *src = 0.f;
}

// Batched weight pre-cast for ALL conv stages in ONE launch (replaces
// n_stages k_cast_wt dispatches).  For each stage s with fp32 master
// W[R][C] (R = KcP rows, C = Cout) at params + w_off[s], emits into the
// shared bf16 scratch buffer:
//   wbuf + bf_off[s]  : bf16 [R][C]        (dgrad fallback B operand)
//   wbuf + bfT_off[s] : bf16 [C][R]        (forward GEMM B operand)
//   wbuf + rot_off[s] : bf16 [Cin][K*K*C]  (implicit dgrad-as-conv B:
//       row ci, col (i'*K+j')*C + c with i' = K-1-i, j' = K-1-j — the
//       180-degree-rotated, channel-transposed kernel; pad rows kc >= Kc
//       have no image here)
struct CastDescs {
  int n;                       // stages (<= 8)
  int R[8], C[8], K[8], Cin[8];
  long long w_off[8], bf_off[8], bfT_off[8], rot_off[8];
  long long p8_off[8];         // 8-padded [C][K*K*8] image (-1 = none;
                               // emitted for Cin < 8 stages so their
                               // forward GEMM can consume the padded
                               // x8 input via the implicit fast path)
  long long cum[9];            // cumulative R*C
};
__global__ void k_cast_wt_all(const float* __restrict__ params,
                              __bf16* __restrict__ wbuf, CastDescs d) {
  const long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= d.cum[d.n]) return;
  int s = 0;
  while (i >= d.cum[s + 1]) ++s;
  const long long q = i - d.cum[s];
  const int C = d.C[s];
  const int kc = (int)(q / C);
  const int c = (int)(q - (long long)kc * C);
  const __bf16 v = (__bf16)params[d.w_off[s] + q];
  wbuf[d.bf_off[s] + q] = v;
  wbuf[d.bfT_off[s] + (long long)c * d.R[s] + kc] = v;
  const int Cin = d.Cin[s], K = d.K[s];
  if (kc < K * K * Cin) {
    const int ci = kc % Cin;
    const int p = kc / Cin;
    const int ki = p / K, kj = p - ki * K;
    const int col = ((K - 1 - ki) * K + (K - 1 - kj)) * C + c;
    wbuf[d.rot_off[s] + (long long)ci * (K * K * C) + col] = v;
    if (d.p8_off[s] >= 0)
      // [C][K*K*8] row per output channel; pad entries (ci >= Cin)
      // stay zero from the buffer's one-time zero init
      wbuf[d.p8_off[s] + (long long)c * (K * K * 8) + p * 8 + ci] = v;
  }
}

// SGD update FUSED with the weight pre-cast: one pass updates every
// parameter (p += step*g; g = 0) and, for conv-weight elements, emits
// the next step's bf16 images (wbf/wbfT/wrot/wp8) from the freshly
// computed value — the separate k_cast_wt_all launch at the next
// forward is skipped (the engine tracks freshness).
__global__ void k_update_cast_all(float* __restrict__ params,
                                  float* __restrict__ grads, long long n,
                                  float step, __bf16* __restrict__ wbuf,
                                  CastDescs d) {
  const long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  const float pnew = params[i] + step * grads[i];
  params[i] = pnew;
  grads[i] = 0.f;
  for (int s = 0; s < d.n; ++s) {
    const long long q = i - d.w_off[s];
    if (q < 0 || q >= d.cum[s + 1] - d.cum[s]) continue;
    const int C = d.C[s];
    const int kc = (int)(q / C);
    const int c = (int)(q - (long long)kc * C);
    const __bf16 v = (__bf16)pnew;
    wbuf[d.bf_off[s] + q] = v;
    wbuf[d.bfT_off[s] + (long long)c * d.R[s] + kc] = v;
    const int Cin = d.Cin[s], K = d.K[s];
    if (kc < K * K * Cin) {
      const int ci = kc % Cin;
      const int p2 = kc / Cin;
      const int ki = p2 / K, kj = p2 - ki * K;
      const int col = ((K - 1 - ki) * K + (K - 1 - kj)) * C + c;
      wbuf[d.rot_off[s] + (long long)ci * (K * K * C) + col] = v;
      if (d.p8_off[s] >= 0)
        wbuf[d.p8_off[s] + (long long)c * (K * K * 8) + p2 * 8 + ci] = v;
    }
    break;
  }
}

// Generic SGD apply + zero:  p += step*g; g = 0  over n params.
__global__ void k_update_n(float* __restrict__ params,
                           float* __restrict__ grads, long long n,
                           float step) {
  const long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) {
    params[i] += step * grads[i];
    grads[i] = 0.f;
  }
}

// MFMA layout self-test: D[16][16] = A[16][32] @ B[32][16], plain fp32 I/O.
__global__ void k_mfma_selftest(const float* __restrict__ A,
                                const float* __restrict__ Bm,
                                float* __restrict__ D) {
  const int lane = threadIdx.x & 63;
  bf16x8 a, b;
#pragma unroll
  for (int r = 0; r < 8; ++r) {
    a[r] = (__bf16)A[(lane & 15) * 32 + (lane >> 4) * 8 + r];
    b[r] = (__bf16)Bm[((lane >> 4) * 8 + r) * 16 + (lane & 15)];
  }
  f32x4 c = {0.f, 0.f, 0.f, 0.f};
  c = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r)
    D[((lane >> 4) * 4 + r) * 16 + (lane & 15)] = c[r];
}

}  // namespace pcnn_deep

// ---------------------------------------------------------------------------
// extern "C" launchers.  act flag: 0 = fp32, 1 = bf16, 2 = fp16.
// ---------------------------------------------------------------------------
using namespace pcnn_deep;

#define PCNN_DISPATCH(flag, ...)                    \
  do {                                               \
    if ((flag) == 1) {                               \
      using act_t = bf16;                            \
      __VA_ARGS__;                                   \
    } else if ((flag) == 2) {                        \
      using act_t = fp16;                            \
      __VA_ARGS__;                                   \
    } else {                                         \
      using act_t = float;                           \
      __VA_ARGS__;                                   \
    }                                                \
  } while (0)

extern "C" {

int pcnn_deep_im2col(const void* x, void* cols, int B, int H, int W, int Cin,
                     int K, int P, int KcP, int actf, void* stream) {
  const long long total = (long long)B * H * W * KcP / 32;  // 32 elems/thread
  dim3 grid((unsigned)((total + 255) / 256)), block(256);
  const bool fast = total < (1LL << 26) && (KcP / 32) <= 4096 && W <= 4096 &&
                    H <= 4096;
  const unsigned long long fd_tpr = fdiv_magic((unsigned)(KcP / 32));
  const unsigned long long fd_w = fdiv_magic((unsigned)W);
  const unsigned long long fd_h = fdiv_magic((unsigned)H);
  if (fast) {
    PCNN_DISPATCH(actf, hipLaunchKernelGGL(
                            (k_im2col<act_t, true>), grid, block, 0,
                            (hipStream_t)stream, (const act_t*)x,
                            (act_t*)cols, B, H, W, Cin, K, P, KcP, fd_tpr,
                            fd_w, fd_h));
  } else {
    PCNN_DISPATCH(actf, hipLaunchKernelGGL(
                            (k_im2col<act_t, false>), grid, block, 0,
                            (hipStream_t)stream, (const act_t*)x,
                            (act_t*)cols, B, H, W, Cin, K, P, KcP, fd_tpr,
                            fd_w, fd_h));
  }
  return (int)hipGetLastError();
}

int pcnn_deep_gemm_ex4(const void* A, const float* Bsrc, const void* Bpre,
                       const float* bias, void* C, long long M, int K, int N,
                       int ldA, int ldC, int b_kxn, int epilogue,
                       const void* imx, int XH, int XW, int XC, int XK,
                       int XP, const void* epi, const float* pw, void* pout,
                       int PK, float* c32, long long c32_cap, int actf,
                       void* stream) {
  const int ntiles = (N + BN - 1) / BN;
  if (imx != nullptr &&
      ((XC % 8) != 0 || M >= (1LL << 26) || XC > 4096 || XK > 4096))
    return -2;  // implicit fast path preconditions (engine falls back)
  if (epilogue == 2 && epi == nullptr) return -3;
  if (epilogue == 3 &&
      (pw == nullptr || pout == nullptr || N > BN || PK < 1 || XW < 1 ||
       64 % XW != 0 || (64 / XW) % PK != 0 || XH % PK != 0 ||
       M % 64 != 0 || (N % 8) != 0))
    return -4;  // fused-pool preconditions (engine falls back)
  const unsigned long long fd_cin = fdiv_magic((unsigned)(XC > 0 ? XC : 1));
  const unsigned long long fd_k = fdiv_magic((unsigned)(XK > 0 ? XK : 1));
  const long long mtiles = (M + BM - 1) / BM;
  const long long mn = mtiles * ntiles;
  // small-K fast path: B resident in LDS, A direct to MFMA registers,
  // pipeline along M (the K-loop never warms at <= 2 iterations)
  if (imx == nullptr && Bpre != nullptr && K <= 128 && (K % 32) == 0 &&
      N <= 64 && (N % 16) == 0 && (M % 64) == 0 && ldA >= K &&
      epilogue != 2) {
    const long long g = mtiles < 1536 ? mtiles : 1536;
    dim3 grid((unsigned)g), block(256);
    PCNN_DISPATCH(actf, hipLaunchKernelGGL(
                            (k_gemm_smallk<act_t>), grid, block, 0,
                            (hipStream_t)stream, (const act_t*)A,
                            (const __bf16*)Bpre, bias, (act_t*)C, M, K, N,
                            ldA, ldC, epilogue, pw, (act_t*)pout, PK, XH,
                            XW));
    return (int)hipGetLastError();
  }
  // split-K policy: a grid under ~3 WGs/CU leaves most of the chip idle
  // AND exposes the per-iteration stage/barrier latency (47 us measured
  // at 64 WGs for 20 us of 1024-WG work); split K until ~1024 WGs.
  int ks_eff = 1, tpc = 0;
  const int ktiles = (K + BKC - 1) / BKC;
  // mode-3 split additionally needs pool windows block-aligned in the
  // combine kernel (256 threads cover whole windows); widths like N=48
  // fail it and simply run unsplit
  const bool epi3_ok =
      epilogue != 3 || 256 % (PK * PK * (N / 8)) == 0;
  if (c32 != nullptr && mn < 768 && ktiles > 1 && (N % 8) == 0 &&
      ldC == N && epi3_ok) {
    int ks = (int)(1024 / mn) + 1;
    if (ks > ktiles) ks = ktiles;
    tpc = (ktiles + ks - 1) / ks;
    ks_eff = (ktiles + tpc - 1) / tpc;
    if ((long long)ks_eff * M * N > c32_cap) {
      ks_eff = 1;  // scratch too small — plain launch
      tpc = 0;
    }
  }
  const bool split = ks_eff > 1;
  dim3 grid((unsigned)(mn * ks_eff)), block(256);
  PCNN_DISPATCH(actf, hipLaunchKernelGGL((k_gemm<act_t, BM>), grid, block,
                                    0, (hipStream_t)stream,
                                    (const act_t*)A, Bsrc,
                                    (const __bf16*)Bpre, bias, (act_t*)C,
                                    M, K, N, ldA, ldC, b_kxn, epilogue,
                                    (const act_t*)imx, XH, XW, XC, XK,
                                    XP, (const act_t*)epi, fd_cin, fd_k,
                                    pw, (act_t*)pout, PK,
                                    split ? c32 : nullptr, tpc));
  if (split) {
    const long long total = (M * N) / 8;
    dim3 g2((unsigned)((total + 255) / 256));
    PCNN_DISPATCH(actf, hipLaunchKernelGGL(
                            (k_split_epi<act_t>), g2, block, 0,
                            (hipStream_t)stream, c32, ks_eff, (act_t*)C, M,
                            N, bias, (const act_t*)epi, epilogue, pw,
                            (act_t*)pout, PK, XH, XW));
  }
  return (int)hipGetLastError();
}

int pcnn_deep_gemm_ex3(const void* A, const float* Bsrc, const void* Bpre,
                       const float* bias, void* C, long long M, int K, int N,
                       int ldA, int ldC, int b_kxn, int epilogue,
                       const void* imx, int XH, int XW, int XC, int XK,
                       int XP, const void* epi, const float* pw, void* pout,
                       int PK, int actf, void* stream) {
  return pcnn_deep_gemm_ex4(A, Bsrc, Bpre, bias, C, M, K, N, ldA, ldC,
                            b_kxn, epilogue, imx, XH, XW, XC, XK, XP, epi,
                            pw, pout, PK, nullptr, 0, actf, stream);
}

int pcnn_deep_gemm_ex(const void* A, const float* Bsrc, const void* Bpre,
                      const float* bias, void* C, long long M, int K, int N,
                      int ldA, int ldC, int b_kxn, int epilogue,
                      const void* imx, int XH, int XW, int XC, int XK,
                      int XP, int actf, void* stream) {
  return pcnn_deep_gemm_ex3(A, Bsrc, Bpre, bias, C, M, K, N, ldA, ldC,
                            b_kxn, epilogue, imx, XH, XW, XC, XK, XP,
                            nullptr, nullptr, nullptr, 0, actf, stream);
}

int pcnn_deep_gemm(const void* A, const float* Bsrc, const float* bias,
                   void* C, long long M, int K, int N, int ldA, int ldC,
                   int b_kxn, int epilogue, int actf, void* stream) {
  return pcnn_deep_gemm_ex(A, Bsrc, nullptr, bias, C, M, K, N, ldA, ldC,
                           b_kxn, epilogue, nullptr, 0, 0, 0, 0, 0, actf,
                           stream);
}

int pcnn_deep_cast_wt(const float* W, void* out, void* outT, int R, int C,
                      void* stream) {
  const long long total = (long long)R * C;
  dim3 grid((unsigned)((total + 255) / 256)), block(256);
  hipLaunchKernelGGL(k_cast_wt, grid, block, 0, (hipStream_t)stream, W,
                     (__bf16*)out, (__bf16*)outT, R, C);
  return (int)hipGetLastError();
}

int pcnn_deep_cast_all(const float* params, void* wbuf, int n_stages,
                       const int* R, const int* C, const int* K,
                       const int* Cin, const long long* w_off,
                       const long long* bf_off, const long long* bfT_off,
                       const long long* rot_off, const long long* p8_off,
                       void* stream) {
  if (n_stages < 1 || n_stages > 8) return -2;
  CastDescs d;
  d.n = n_stages;
  d.cum[0] = 0;
  for (int s = 0; s < n_stages; ++s) {
    d.R[s] = R[s];
    d.C[s] = C[s];
    d.K[s] = K[s];
    d.Cin[s] = Cin[s];
    d.w_off[s] = w_off[s];
    d.bf_off[s] = bf_off[s];
    d.bfT_off[s] = bfT_off[s];
    d.rot_off[s] = rot_off[s];
    d.p8_off[s] = p8_off ? p8_off[s] : -1;
    d.cum[s + 1] = d.cum[s] + (long long)R[s] * C[s];
  }
  dim3 grid((unsigned)((d.cum[n_stages] + 255) / 256)), block(256);
  hipLaunchKernelGGL(k_cast_wt_all, grid, block, 0, (hipStream_t)stream,
                     params, (__bf16*)wbuf, d);
  return (int)hipGetLastError();
}

int pcnn_deep_update_cast(float* params, float* grads, long long n,
                          float step, void* wbuf, int n_stages, const int* R,
                          const int* C, const int* K, const int* Cin,
                          const long long* w_off, const long long* bf_off,
                          const long long* bfT_off, const long long* rot_off,
                          const long long* p8_off, void* stream) {
  if (n_stages < 1 || n_stages > 8) return -2;
  CastDescs d;
  d.n = n_stages;
  d.cum[0] = 0;
  for (int s2 = 0; s2 < n_stages; ++s2) {
    d.R[s2] = R[s2];
    d.C[s2] = C[s2];
    d.K[s2] = K[s2];
    d.Cin[s2] = Cin[s2];
    d.w_off[s2] = w_off[s2];
    d.bf_off[s2] = bf_off[s2];
    d.bfT_off[s2] = bfT_off[s2];
    d.rot_off[s2] = rot_off[s2];
    d.p8_off[s2] = p8_off ? p8_off[s2] : -1;
    d.cum[s2 + 1] = d.cum[s2] + (long long)R[s2] * C[s2];
  }
  dim3 grid((unsigned)((n + 255) / 256)), block(256);
  hipLaunchKernelGGL(k_update_cast_all, grid, block, 0,
                     (hipStream_t)stream, params, grads, n, step,
                     (__bf16*)wbuf, d);
  return (int)hipGetLastError();
}

int pcnn_deep_wgrad_gemm_ex2(const void* cols, const void* dpre, float* dW,
                             float* part, long long M, int KcP, int N,
                             int MS, const void* imx, int XH, int XW, int XC,
                             int XK, int XP, float* db, int actf,
                             void* stream) {
  if (imx != nullptr && ((XC % 8) != 0 || M >= (1LL << 26) || XC > 4096 ||
                         XK > 4096 || XW > 4096 || XH > 4096))
    return -2;  // implicit fast path preconditions (engine falls back)
  const unsigned long long fd_cin = fdiv_magic((unsigned)(XC > 0 ? XC : 1));
  const unsigned long long fd_k = fdiv_magic((unsigned)(XK > 0 ? XK : 1));
  const unsigned long long fd_xw = fdiv_magic((unsigned)(XW > 0 ? XW : 1));
  const unsigned long long fd_xh = fdiv_magic((unsigned)(XH > 0 ? XH : 1));
  const int ktiles = (KcP + BM - 1) / BM;
  const int ntiles = (N + BN - 1) / BN;
  dim3 grid((unsigned)(ktiles * ntiles * MS)), block(256);
  PCNN_DISPATCH(actf, hipLaunchKernelGGL((k_wgrad_gemm<act_t>), grid, block,
                                          0, (hipStream_t)stream,
                                          (const act_t*)cols,
                                          (const act_t*)dpre, dW, part, M, KcP,
                                          N, MS, (const act_t*)imx, XH, XW, XC,
                                          XK, XP, db, fd_cin, fd_k, fd_xw,
                                          fd_xh));
  if (part) {
    const long long total = (long long)KcP * N;
    dim3 g2((unsigned)((total / 8 + 255) / 256));
    hipLaunchKernelGGL(k_wsum, g2, block, 0, (hipStream_t)stream, part, dW,
                       total, MS);
  }
  return (int)hipGetLastError();
}

int pcnn_deep_wgrad_gemm_ex(const void* cols, const void* dpre, float* dW,
                            float* part, long long M, int KcP, int N, int MS,
                            const void* imx, int XH, int XW, int XC, int XK,
                            int XP, int actf, void* stream) {
  return pcnn_deep_wgrad_gemm_ex2(cols, dpre, dW, part, M, KcP, N, MS, imx,
                                  XH, XW, XC, XK, XP, nullptr, actf, stream);
}

// n_stages wgrad GEMMs in one launch (see k_wgrad_multi).  a[s] is the
// materialized cols when implicit[s]==0, else the NHWC source tensor.
int pcnn_deep_wgrad_multi(int n_stages, const void* const* a,
                          const void* const* dpre, float* const* dW,
                          float* const* db, const long long* M,
                          const int* KcP, const int* N, const int* MS,
                          const int* implicit, const int* XH, const int* XW,
                          const int* XC, const int* XK, const int* XP,
                          int actf, void* stream) {
  if (n_stages < 1 || n_stages > 8) return -2;
  WgDesc d;
  d.n = n_stages;
  d.blk_cum[0] = 0;
  for (int s = 0; s < n_stages; ++s) {
    if (implicit[s] &&
        ((XC[s] % 8) != 0 || M[s] >= (1LL << 26) || XC[s] > 4096 ||
         XK[s] > 4096 || XW[s] > 4096 || XH[s] > 4096))
      return -3;
    d.a[s] = a[s];
    d.dpre[s] = dpre[s];
    d.dW[s] = dW[s];
    d.db[s] = db[s];
    d.M[s] = M[s];
    d.KcP[s] = KcP[s];
    d.N[s] = N[s];
    d.MS[s] = MS[s];
    d.implicit[s] = implicit[s];
    d.XH[s] = XH[s];
    d.XW[s] = XW[s];
    d.XC[s] = XC[s];
    d.XK[s] = XK[s];
    d.XP[s] = XP[s];
    d.fd_cin[s] = fdiv_magic((unsigned)(XC[s] > 0 ? XC[s] : 1));
    d.fd_k[s] = fdiv_magic((unsigned)(XK[s] > 0 ? XK[s] : 1));
    d.fd_xw[s] = fdiv_magic((unsigned)(XW[s] > 0 ? XW[s] : 1));
    d.fd_xh[s] = fdiv_magic((unsigned)(XH[s] > 0 ? XH[s] : 1));
    const long long blocks = (long long)((KcP[s] + BM - 1) / BM) *
                             ((N[s] + BN - 1) / BN) * MS[s];
    d.blk_cum[s + 1] = d.blk_cum[s] + blocks;
  }
  dim3 grid((unsigned)d.blk_cum[n_stages]), block(256);
  PCNN_DISPATCH(actf, hipLaunchKernelGGL((k_wgrad_multi<act_t>), grid,
                                          block, 0, (hipStream_t)stream,
                                          d));
  return (int)hipGetLastError();
}

int pcnn_deep_pad_channels(const void* x, void* x8, long long npix, int Cin,
                           int actf, void* stream) {
  if (Cin < 1 || Cin > 8) return -2;
  dim3 grid((unsigned)((npix + 255) / 256)), block(256);
  PCNN_DISPATCH(actf, hipLaunchKernelGGL((k_pad_channels<act_t>), grid,
                                          block, 0, (hipStream_t)stream,
                                          (const act_t*)x, (act_t*)x8, npix,
                                          Cin));
  return (int)hipGetLastError();
}

int pcnn_deep_remap_dw8(float* dW8, float* dW, int KK, int Cin, int Cout,
                        void* stream) {
  const int total = KK * Cin * Cout;
  dim3 grid((unsigned)((total + 255) / 256)), block(256);
  hipLaunchKernelGGL(k_remap_dw8, grid, block, 0, (hipStream_t)stream, dW8,
                     dW, KK, Cin, Cout);
  return (int)hipGetLastError();
}

int pcnn_deep_wgrad_gemm(const void* cols, const void* dpre, float* dW,
                         float* part, long long M, int KcP, int N, int MS,
                         int actf, void* stream) {
  return pcnn_deep_wgrad_gemm_ex(cols, dpre, dW, part, M, KcP, N, MS, nullptr,
                                 0, 0, 0, 0, 0, actf, stream);
}

int pcnn_deep_colsum(const void* dpre, float* part, float* db, long long M,
                     int N, int slices, int actf, void* stream) {
  dim3 grid((unsigned)slices), block(256);
  PCNN_DISPATCH(actf, hipLaunchKernelGGL((k_colsum<act_t>), grid, block, 0,
                                          (hipStream_t)stream,
                                          (const act_t*)dpre, part, M, N,
                                          slices));
  hipLaunchKernelGGL(k_colsum_fin, dim3(1), dim3(256), 0,
                     (hipStream_t)stream, part, db, N, slices);
  return (int)hipGetLastError();
}

int pcnn_deep_col2im_sigbwd(const void* dcols, const void* pout, void* out,
                            int B, int H, int W, int Cin, int K, int P,
                            int KcP, int actf, void* stream) {
  const long long total = (long long)B * H * W * Cin / 8;
  dim3 grid((unsigned)((total + 255) / 256)), block(256);
  const bool fast = total < (1LL << 26) && (Cin / 8) <= 4096 && W <= 4096 &&
                    H <= 4096;
  const unsigned long long fd_cpr = fdiv_magic((unsigned)(Cin / 8));
  const unsigned long long fd_w = fdiv_magic((unsigned)W);
  const unsigned long long fd_h = fdiv_magic((unsigned)H);
  if (fast) {
    PCNN_DISPATCH(actf, hipLaunchKernelGGL((k_col2im_sigbwd<act_t, true>),
                                            grid, block, 0,
                                            (hipStream_t)stream,
                                            (const act_t*)dcols,
                                            (const act_t*)pout, (act_t*)out,
                                            B, H, W, Cin, K, P, KcP, fd_cpr,
                                            fd_w, fd_h));
  } else {
    PCNN_DISPATCH(actf, hipLaunchKernelGGL((k_col2im_sigbwd<act_t, false>),
                                            grid, block, 0,
                                            (hipStream_t)stream,
                                            (const act_t*)dcols,
                                            (const act_t*)pout, (act_t*)out,
                                            B, H, W, Cin, K, P, KcP, fd_cpr,
                                            fd_w, fd_h));
  }
  return (int)hipGetLastError();
}

int pcnn_deep_pool_fwd(const void* a, const float* pw, void* pout, int B,
                       int H, int W, int C, int K, int actf, void* stream) {
  const long long total = (long long)B * (H / K) * (W / K) * C / 8;
  dim3 grid((unsigned)((total + 255) / 256)), block(256);
  PCNN_DISPATCH(actf, hipLaunchKernelGGL((k_pool_fwd<act_t>), grid, block, 0,
                                          (hipStream_t)stream,
                                          (const act_t*)a, pw, (act_t*)pout,
                                          B, H, W, C, K));
  return (int)hipGetLastError();
}

int pcnn_deep_pool_bwd(const void* dppre, const void* a, const float* pw,
                       void* dapre, int B, int H, int W, int C, int K,
                       int actf, void* stream) {
  const long long total = (long long)B * H * W * C / 8;
  dim3 grid((unsigned)((total + 255) / 256)), block(256);
  PCNN_DISPATCH(actf, hipLaunchKernelGGL((k_pool_bwd<act_t>), grid, block, 0,
                                          (hipStream_t)stream,
                                          (const act_t*)dppre,
                                          (const act_t*)a, pw, (act_t*)dapre,
                                          B, H, W, C, K));
  return (int)hipGetLastError();
}

int pcnn_deep_pool_wgrad(const void* dppre, const void* a, float* dpw, int B,
                         int H, int W, int C, int K, int G, int actf,
                         void* stream) {
  dim3 grid(G), block(256);
  if (C % 8 == 0 && K == 2) {
    // vectorized path: 16B ld8v over 8 consecutive channels per thread.
    // (K=3/4 instantiations need 200+ VGPRs with the 2-in-flight pipeline
    // — occupancy 1 — so larger pools keep the scalar kernel.)
    PCNN_DISPATCH(actf, hipLaunchKernelGGL((k_pool_wgrad8<act_t, 2>), grid,
                                            block, 0, (hipStream_t)stream,
                                            (const act_t*)dppre,
                                            (const act_t*)a, dpw, B, H, W, C,
                                            G));
    return (int)hipGetLastError();
  }
  PCNN_DISPATCH(actf, hipLaunchKernelGGL((k_pool_wgrad<act_t>), grid, block,
                                          0, (hipStream_t)stream,
                                          (const act_t*)dppre,
                                          (const act_t*)a, dpw, B, H, W, C, K,
                                          G));
  return (int)hipGetLastError();
}

int pcnn_deep_pool_wbwd(const void* dppre, const void* a, const float* pw,
                        void* dapre, float* dpw, int B, int H, int W, int C,
                        int K, int G, int actf, void* stream) {
  if (C % 8 != 0 || K != 2) return -2;  // engine falls back to 2 kernels
  dim3 grid(G), block(256);
  PCNN_DISPATCH(actf, hipLaunchKernelGGL((k_pool_wbwd8<act_t, 2>), grid,
                                          block, 0, (hipStream_t)stream,
                                          (const act_t*)dppre,
                                          (const act_t*)a, pw,
                                          (act_t*)dapre, dpw, B, H, W, C,
                                          G));
  return (int)hipGetLastError();
}

int pcnn_deep_fc_fwd2(const void* flat, const float* fw, const float* fb,
                      const int* labels, float* yg, float* dzg,
                      float* loss_accum, int* correct_accum, int B, int FCIN,
                      int NCLS, int mode, void* dflat, int actf,
                      void* stream) {
  dim3 grid(B), block(256);
  PCNN_DISPATCH(actf, hipLaunchKernelGGL((k_fc_fwd<act_t>), grid, block, 0,
                                          (hipStream_t)stream,
                                          (const act_t*)flat, fw, fb, labels,
                                          yg, dzg, loss_accum, correct_accum,
                                          B, FCIN, NCLS, mode,
                                          (act_t*)dflat));
  return (int)hipGetLastError();
}

int pcnn_deep_fc_fwd(const void* flat, const float* fw, const float* fb,
                     const int* labels, float* yg, float* dzg,
                     float* loss_accum, int* correct_accum, int B, int FCIN,
                     int NCLS, int mode, int actf, void* stream) {
  return pcnn_deep_fc_fwd2(flat, fw, fb, labels, yg, dzg, loss_accum,
                           correct_accum, B, FCIN, NCLS, mode, nullptr,
                           actf, stream);
}

int pcnn_deep_fc_bwd(const float* dzg, const void* flat, const float* fw,
                     void* dflat, int B, int FCIN, int NCLS, int actf,
                     void* stream) {
  const long long total = (long long)B * FCIN;
  dim3 grid((unsigned)((total + 255) / 256)), block(256);
  PCNN_DISPATCH(actf, hipLaunchKernelGGL((k_fc_bwd<act_t>), grid, block, 0,
                                          (hipStream_t)stream, dzg,
                                          (const act_t*)flat, fw,
                                          (act_t*)dflat, B, FCIN, NCLS));
  return (int)hipGetLastError();
}

int pcnn_deep_fc_wgrad(const float* dzg, const void* flat, float* gfw,
                       float* gfb, int B, int FCIN, int NCLS, int FS,
                       int actf, void* stream) {
  const int nw = NCLS * FCIN;
  const int blocks_per_slice = (nw + 255 + NCLS) / 256 + 1;
  dim3 grid(blocks_per_slice * FS), block(256);
  PCNN_DISPATCH(actf, hipLaunchKernelGGL((k_fc_wgrad<act_t>), grid, block, 0,
                                          (hipStream_t)stream, dzg,
                                          (const act_t*)flat, gfw, gfb, B,
                                          FCIN, NCLS, FS));
  return (int)hipGetLastError();
}

int pcnn_deep_update(float* params, float* grads, long long n, float step,
                     void* stream) {
  dim3 grid((unsigned)((n + 255) / 256)), block(256);
  hipLaunchKernelGGL(k_update_n, grid, block, 0, (hipStream_t)stream, params,
                     grads, n, step);
  return (int)hipGetLastError();
}

int pcnn_deep_mfma_selftest(const float* A, const float* Bm, float* D,
                            void* stream) {
  hipLaunchKernelGGL(k_mfma_selftest, dim3(1), dim3(64), 0,
                     (hipStream_t)stream, A, Bm, D);
  return (int)hipGetLastError();
}
}
