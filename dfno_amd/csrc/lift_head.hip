// gfx950 fused FNO lift head: y = gelu(W2 @ gelu(W1 @t x + b1) + b2).
//
// The reference runs this as a time-lift einsum (T_in -> T_out along the
// trailing dim), a GELU pass, a channel-lift einsum (C_in -> width) and
// another GELU (/root/reference/dfno/dfno.py:310-311,333-338).  rocBLAS
// handles the T_in = 1 time lift (an outer product, M ~ 5e5, K = 1)
// pathologically (~1.25 ms for a 250 MB write), and the unfused chain
// costs ~4 ms/step at the flagship config.  Here the whole lift is one
// pass: read the (tiny) input column, produce the [width, T_out] output
// column per grid point.  Restricted to T_in == 1 (the two-phase flagship;
// other configs use the unfused path).
//
// fp32/bf16 forward+backward run the lane-per-k variants: each 64-lane
// wave covers TWO s-points with one time column per lane, so every out
// write / gy read is a contiguous Tn-dword run; backward's gW2/gb2
// reduce as v_mfma_f32_16x16x4 over per-wave LDS tiles (ones-column
// trick for gb2) with fragments carried across pairs, and gW1/gb1
// accumulate per-lane (k is fixed).  The chunked per-thread kernels
// below them serve fp64/odd shapes.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <torch/extension.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>

#include "kernels.h"
#include "gelu_math.h"

namespace {

constexpr int kBlock = 256;

template <typename T>
__device__ __forceinline__ T gelu_(T z) { return dfno_gelu::gelu(z); }

template <typename T>
__device__ __forceinline__ T gelu_g_(T z) { return dfno_gelu::gelu_grad(z); }

__device__ __forceinline__ float lh_ld(const float* p) { return *p; }
__device__ __forceinline__ float lh_ld(const unsigned short* p) {
  return __uint_as_float(((unsigned int)*p) << 16);
}
__device__ __forceinline__ void lh_st(float* p, float v) { *p = v; }
__device__ __forceinline__ void lh_st(unsigned short* p, float v) {
  __hip_bfloat16 h = __float2bfloat16(v);
  *p = *reinterpret_cast<unsigned short*>(&h);
}

template <typename T>
__device__ __forceinline__ T lh_wave_sum(T v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}

// x: [B, C, S] (T_in == 1 folded away), out: [B, W, S, T]
// W1: [T, 1], b1: [T], W2: [W, C], b2: [W]
//
// Both kernels chunk the time axis (KC columns at a time): the first cut
// held h/z1/gh[CCAP][TCAP] per thread (384 floats -> 256 VGPRs + ~600
// spilled, 1 wave/SIMD, every element round-tripping scratch).  Chunked,
// the live set is ~2*CCAP*KC + temps and z1 is recomputed (2 FMA) where
// its gradient factor is needed.
template <typename T, int CCAP, int TCAP, int WCAP, int WT = 0>
__global__ __launch_bounds__(kBlock, 3) void lift_head_fwd_kernel(
    const T* __restrict__ x, const T* __restrict__ W1, const T* __restrict__ b1,
    const T* __restrict__ W2, const T* __restrict__ b2, T* __restrict__ out,
    int B, int C, int W, int Tn, long S) {
  constexpr int KC = 10;
  __shared__ T w1[TCAP], bb1[TCAP], w2[WCAP * CCAP], bb2[WCAP];
  for (int k = threadIdx.x; k < Tn; k += kBlock) { w1[k] = W1[k]; bb1[k] = b1[k]; }
  for (int k = threadIdx.x; k < W * C; k += kBlock) w2[k] = W2[k];
  for (int k = threadIdx.x; k < W; k += kBlock) bb2[k] = b2[k];
  __syncthreads();

  long t0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long t = t0; t < (long)B * S; t += stride) {
    long b = t / S;
    long s = t % S;

    T xv[CCAP];
#pragma unroll
    for (int c = 0; c < CCAP; ++c)
      if (c < C) xv[c] = x[(b * C + c) * S + s];

    for (int k0 = 0; k0 < Tn; k0 += KC) {
      const int kn = min(KC, Tn - k0);
      // h[c][k] = gelu(W1[k] * x[c] + b1[k]) for this chunk
      T h[CCAP][KC];
#pragma unroll
      for (int c = 0; c < CCAP; ++c) {
        if (c < C) {
#pragma unroll
          for (int k = 0; k < KC; ++k)
            if (k < kn) h[c][k] = gelu_(w1[k0 + k] * xv[c] + bb1[k0 + k]);
        }
      }
#pragma unroll 4
      for (int w = 0; w < (WT > 0 ? WT : 512); ++w) {
        if (WT == 0 && w >= W) break;
        T acc[KC];
#pragma unroll
        for (int k = 0; k < KC; ++k)
          if (k < kn) acc[k] = bb2[w];
#pragma unroll
        for (int c = 0; c < CCAP; ++c) {
          if (c < C) {
            T wv = w2[(size_t)w * C + c];
#pragma unroll
            for (int k = 0; k < KC; ++k)
              if (k < kn) acc[k] += wv * h[c][k];
          }
        }
        T* dst = out + ((b * W + w) * S + s) * Tn + k0;
#pragma unroll
        for (int k = 0; k < KC; ++k)
          if (k < kn) dst[k] = gelu_(acc[k]);
      }
    }
  }
}

// gy: [B, W, S, T]; outputs gx [B, C, S] and gW1/gb1/gW2/gb2 via atomics.
template <typename T, int CCAP, int TCAP, int WCAP, int WT = 0>
__global__ __launch_bounds__(kBlock, 3) void lift_head_bwd_kernel(
    const T* __restrict__ gy, const T* __restrict__ x,
    const T* __restrict__ W1, const T* __restrict__ b1,
    const T* __restrict__ W2, const T* __restrict__ b2,
    T* __restrict__ gx, T* __restrict__ gW1, T* __restrict__ gb1,
    T* __restrict__ gW2, T* __restrict__ gb2,
    int B, int C, int W, int Tn, long S) {
  constexpr int KC = 10;
  __shared__ T w1[TCAP], bb1[TCAP], w2[WCAP * CCAP], bb2[WCAP];
  // per-wave partial accumulators
  __shared__ T a_gW1[4][TCAP], a_gb1[4][TCAP], a_gW2[4][WCAP * CCAP], a_gb2[4][WCAP];
  for (int k = threadIdx.x; k < Tn; k += kBlock) { w1[k] = W1[k]; bb1[k] = b1[k]; }
  for (int k = threadIdx.x; k < W * C; k += kBlock) w2[k] = W2[k];
  for (int k = threadIdx.x; k < W; k += kBlock) bb2[k] = b2[k];
  for (int k = threadIdx.x; k < 4 * TCAP; k += kBlock)
    a_gW1[k / TCAP][k % TCAP] = a_gb1[k / TCAP][k % TCAP] = T(0);
  for (int k = threadIdx.x; k < 4 * WCAP * CCAP; k += kBlock)
    a_gW2[k / (WCAP * CCAP)][k % (WCAP * CCAP)] = T(0);
  for (int k = threadIdx.x; k < 4 * WCAP; k += kBlock)
    a_gb2[k / WCAP][k % WCAP] = T(0);
  __syncthreads();

  const int wave = threadIdx.x / 64;
  const int lane = threadIdx.x % 64;

  long t0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long t = t0; t < (long)B * S; t += stride) {
    long b = t / S;
    long s = t % S;

    T xv[CCAP], gxa[CCAP];
#pragma unroll
    for (int c = 0; c < CCAP; ++c) {
      if (c < C) {
        xv[c] = x[(b * C + c) * S + s];
        gxa[c] = T(0);
      }
    }

    for (int k0 = 0; k0 < Tn; k0 += KC) {
      const int kn = min(KC, Tn - k0);
      T h[CCAP][KC], gh[CCAP][KC];
#pragma unroll
      for (int c = 0; c < CCAP; ++c) {
        if (c < C) {
#pragma unroll
          for (int k = 0; k < KC; ++k) {
            if (k < kn) {
              h[c][k] = gelu_(w1[k0 + k] * xv[c] + bb1[k0 + k]);
              gh[c][k] = T(0);
            }
          }
        }
      }

#pragma unroll 4
      for (int w = 0; w < (WT > 0 ? WT : 512); ++w) {
        if (WT == 0 && w >= W) break;
        const T* gyp = gy + ((b * W + w) * S + s) * Tn + k0;
        T gz2[KC];
        T gb2p = T(0);
#pragma unroll
        for (int k = 0; k < KC; ++k) {
          if (k < kn) {
            // recompute z2[w][k]
            T z2 = bb2[w];
#pragma unroll
            for (int c = 0; c < CCAP; ++c)
              if (c < C) z2 += w2[(size_t)w * C + c] * h[c][k];
            gz2[k] = gyp[k] * gelu_g_(z2);
            gb2p += gz2[k];
          }
        }
        // accumulate gh and wave-reduced gW2 partials
#pragma unroll
        for (int c = 0; c < CCAP; ++c) {
          if (c < C) {
            T wv = w2[(size_t)w * C + c];
            T gw2p = T(0);
#pragma unroll
            for (int k = 0; k < KC; ++k) {
              if (k < kn) {
                gh[c][k] += wv * gz2[k];
                gw2p += gz2[k] * h[c][k];
              }
            }
            gw2p = lh_wave_sum(gw2p);
            if (lane == 0) a_gW2[wave][w * C + c] += gw2p;
          }
        }
        gb2p = lh_wave_sum(gb2p);
        if (lane == 0) a_gb2[wave][w] += gb2p;
      }

      // gz1 = gh * gelu'(z1); gx[c] += sum_k W1[k] gz1[c][k];
      // gW1[k] += sum_c gz1[c][k] * x[c]; gb1[k] += sum_c gz1[c][k]
#pragma unroll
      for (int c = 0; c < CCAP; ++c) {
        if (c < C) {
#pragma unroll
          for (int k = 0; k < KC; ++k) {
            if (k < kn) {
              T gz1 = gh[c][k] * gelu_g_(w1[k0 + k] * xv[c] + bb1[k0 + k]);
              gh[c][k] = gz1;  // reuse as gz1 for the reductions below
              gxa[c] += w1[k0 + k] * gz1;
            }
          }
        }
      }
#pragma unroll
      for (int k = 0; k < KC; ++k) {
        if (k < kn) {
          T pw = T(0), pb = T(0);
#pragma unroll
          for (int c = 0; c < CCAP; ++c) {
            if (c < C) {
              pw += gh[c][k] * xv[c];
              pb += gh[c][k];
            }
          }
          pw = lh_wave_sum(pw);
          pb = lh_wave_sum(pb);
          if (lane == 0) { a_gW1[wave][k0 + k] += pw; a_gb1[wave][k0 + k] += pb; }
        }
      }
    }

#pragma unroll
    for (int c = 0; c < CCAP; ++c)
      if (c < C) gx[(b * C + c) * S + s] = gxa[c];
  }

  __syncthreads();
  for (int k = threadIdx.x; k < Tn; k += kBlock) {
    T v = a_gW1[0][k] + a_gW1[1][k] + a_gW1[2][k] + a_gW1[3][k];
    if (v != T(0)) atomicAdd(&gW1[k], v);
    v = a_gb1[0][k] + a_gb1[1][k] + a_gb1[2][k] + a_gb1[3][k];
    if (v != T(0)) atomicAdd(&gb1[k], v);
  }
  for (int k = threadIdx.x; k < W * C; k += kBlock) {
    T v = a_gW2[0][k] + a_gW2[1][k] + a_gW2[2][k] + a_gW2[3][k];
    if (v != T(0)) atomicAdd(&gW2[k], v);
  }
  for (int k = threadIdx.x; k < W; k += kBlock) {
    T v = a_gb2[0][k] + a_gb2[1][k] + a_gb2[2][k] + a_gb2[3][k];
    if (v != T(0)) atomicAdd(&gb2[k], v);
  }
}

// fp32 lane-per-k variants: each 64-lane wave covers TWO s-points with
// lanes 0..Tn-1 / 32..32+Tn-1 owning one time column each, so every gy
// read and out write is a contiguous Tn-dword run (the per-thread layout
// above streams 30-float runs per lane -> fully scattered wave accesses;
// measured 699/843 us vs ~110/130 us of traffic).  Per-lane state is
// ~40 VGPRs; gW1/gb1 partials accumulate in registers (k fixed per lane).
template <int WT, typename TIO = float>
__global__ __launch_bounds__(kBlock) void lift_head_fwd_lk_kernel(
    const TIO* __restrict__ x, const float* __restrict__ W1,
    const float* __restrict__ b1, const float* __restrict__ W2,
    const float* __restrict__ b2, TIO* __restrict__ out,
    int B, int C, int W, int Tn, long S) {
  constexpr int CC = 4;
  __shared__ float w1[32], bb1[32], w2[24 * CC], bb2[24];
  for (int k = threadIdx.x; k < Tn; k += kBlock) { w1[k] = W1[k]; bb1[k] = b1[k]; }
  for (int k = threadIdx.x; k < W * C; k += kBlock) w2[k] = W2[k];
  for (int k = threadIdx.x; k < W; k += kBlock) bb2[k] = b2[k];
  __syncthreads();

  const int lane = (int)(threadIdx.x & 63);
  const int half = lane >> 5;                 // 0 or 1: which s of the pair
  const int k = lane & 31;
  const bool kv = k < Tn;
  const long npair = ((long)B * S + 1) / 2;
  long t0 = (long)blockIdx.x * (blockDim.x / 64) + (threadIdx.x >> 6);
  long stride = (long)gridDim.x * (blockDim.x / 64);
  for (long t = t0; t < npair; t += stride) {
    const long e = 2 * t + half;              // element index
    const bool ev = e < (long)B * S;
    const long b = e / S;
    const long s = e - b * S;

    float h[CC];
#pragma unroll
    for (int c = 0; c < CC; ++c) {
      if (c < C) {
        const float xv = ev ? lh_ld(x + (b * C + c) * S + s) : 0.f;
        h[c] = kv ? gelu_(w1[k] * xv + bb1[k]) : 0.f;
      }
    }
#pragma unroll 4
    for (int w = 0; w < (WT > 0 ? WT : 512); ++w) {
      if (WT == 0 && w >= W) break;
      float acc = bb2[w];
#pragma unroll
      for (int c = 0; c < CC; ++c)
        if (c < C) acc += w2[w * C + c] * h[c];
      if (ev && kv)
        lh_st(out + ((b * W + w) * S + s) * Tn + k, gelu_(acc));
    }
  }
}

typedef float f32x4_lh __attribute__((ext_vector_type(4)));

template <int WT, typename TIO = float>
__global__ __launch_bounds__(kBlock) void lift_head_bwd_lk_kernel(
    const TIO* __restrict__ gy, const TIO* __restrict__ x,
    const float* __restrict__ W1, const float* __restrict__ b1,
    const float* __restrict__ W2, const float* __restrict__ b2,
    TIO* __restrict__ gx, float* __restrict__ gW1, float* __restrict__ gb1,
    float* __restrict__ gW2, float* __restrict__ gb2,
    int B, int C, int W, int Tn, long S) {
  constexpr int CC = 4;
  constexpr int GLD = 68;                     // gz/h tile row pad (banks)
  __shared__ float w1[32], bb1[32], w2[24 * CC], bb2[24];
  // per-wave tiles: gz [24 w][64 lanes] and h [4 c][64 lanes]; the gW2 and
  // gb2 reductions run as v_mfma_f32_16x16x4 over these (B-operand column
  // 4 is a constant ones column giving gb2) with the C fragments carried
  // in registers across pairs — the per-pair 6-shuffle wave_sum chains of
  // the first cut were the whole kernel's latency.
  __shared__ float gzt[4][24 * GLD];
  __shared__ float ht[4][CC * GLD];
  for (int k = threadIdx.x; k < Tn; k += kBlock) { w1[k] = W1[k]; bb1[k] = b1[k]; }
  for (int k = threadIdx.x; k < W * C; k += kBlock) w2[k] = W2[k];
  for (int k = threadIdx.x; k < W; k += kBlock) bb2[k] = b2[k];
  __syncthreads();

  const int wave = (int)(threadIdx.x >> 6);
  const int lane = (int)(threadIdx.x & 63);
  const int half = lane >> 5;
  const int k = lane & 31;
  const bool kv = k < Tn;
  const int l16 = lane & 15;
  const int kg = lane >> 4;
  float pgW1 = 0.f, pgb1 = 0.f;               // per-lane: k is fixed
  f32x4_lh wacc[2];                           // gW2/gb2 frags (2 m-tiles)
  wacc[0] = f32x4_lh{0.f, 0.f, 0.f, 0.f};
  wacc[1] = f32x4_lh{0.f, 0.f, 0.f, 0.f};
  float* gzw = &gzt[wave][0];
  float* hw = &ht[wave][0];

  const long npair = ((long)B * S + 1) / 2;
  long t0 = (long)blockIdx.x * (blockDim.x / 64) + (threadIdx.x >> 6);
  long stride = (long)gridDim.x * (blockDim.x / 64);
  for (long t = t0; t < npair; t += stride) {
    const long e = 2 * t + half;
    const bool ev = e < (long)B * S;
    const long b = e / S;
    const long s = e - b * S;

    float xv[CC], h[CC], gh[CC];
#pragma unroll
    for (int c = 0; c < CC; ++c) {
      if (c < C) {
        xv[c] = ev ? lh_ld(x + (b * C + c) * S + s) : 0.f;
        h[c] = kv ? gelu_(w1[k] * xv[c] + bb1[k]) : 0.f;
        gh[c] = 0.f;
        hw[c * GLD + lane] = h[c];
      }
    }
    // all W gy columns preloaded back-to-back: one memory round trip per
    // pair instead of one per unroll batch (the loads were the kernel's
    // whole latency — all reduction variants measured the same ~900 us)
    float gyv[WT > 0 ? WT : 24];
#pragma unroll
    for (int w = 0; w < (WT > 0 ? WT : 24); ++w) {
      gyv[w] = (w < W && ev && kv)
                   ? lh_ld(gy + ((b * W + w) * S + s) * Tn + k) : 0.f;
    }
#pragma unroll 4
    for (int w = 0; w < (WT > 0 ? WT : 512); ++w) {
      if (WT == 0 && w >= W) break;
      float z2 = bb2[w];
#pragma unroll
      for (int c = 0; c < CC; ++c)
        if (c < C) z2 += w2[w * C + c] * h[c];
      const float gz2 = gyv[w] * gelu_g_(z2);
      gzw[w * GLD + lane] = gz2;
#pragma unroll
      for (int c = 0; c < CC; ++c)
        if (c < C) gh[c] += w2[w * C + c] * gz2;
    }
    // gW2/gb2 fragment update from this pair's tiles (wave-synchronous:
    // same wave wrote gzt/ht just above)
#pragma unroll
    for (int mt = 0; mt < 2; ++mt) {
      const int wrow = mt * 16 + l16;
      const bool av = wrow < W;
#pragma unroll 4
      for (int k0 = 0; k0 < 64; k0 += 4) {
        const float a = av ? gzw[wrow * GLD + k0 + kg] : 0.f;
        const float bb = (l16 < C) ? hw[l16 * GLD + k0 + kg]
                                   : (l16 == C ? 1.f : 0.f);
        wacc[mt] = __builtin_amdgcn_mfma_f32_16x16x4f32(a, bb, wacc[mt],
                                                        0, 0, 0);
      }
    }
    // gz1 = gh * gelu'(z1); per-lane gW1/gb1 partials; gx via 32-lane
    // segmented reduce (k lanes of each s-half)
#pragma unroll
    for (int c = 0; c < CC; ++c) {
      if (c < C) {
        const float gz1 = kv ? gh[c] * gelu_g_(w1[k] * xv[c] + bb1[k]) : 0.f;
        pgW1 += gz1 * xv[c];
        pgb1 += gz1;
        float gxa = kv ? w1[k] * gz1 : 0.f;
#pragma unroll
        for (int off = 16; off > 0; off >>= 1)
          gxa += __shfl_xor(gxa, off, 64);
        if (ev && k == 0) lh_st(gx + (b * C + c) * S + s, gxa);
      }
    }
  }

  // flush: pair the two s-halves for gW1/gb1, then one atomic per (wave, k);
  // gW2/gb2 from the MFMA fragments (D[m=w][n]: n<C -> gW2, n==C -> gb2)
  pgW1 += __shfl_xor(pgW1, 32, 64);
  pgb1 += __shfl_xor(pgb1, 32, 64);
  if (half == 0 && kv) {
    atomicAdd(&gW1[k], pgW1);
    atomicAdd(&gb1[k], pgb1);
  }
#pragma unroll
  for (int mt = 0; mt < 2; ++mt) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int w = mt * 16 + kg * 4 + r;
      const float v = wacc[mt][r];
      if (w < W && v != 0.f) {
        if (l16 < C) atomicAdd(&gW2[w * C + l16], v);
        else if (l16 == C) atomicAdd(&gb2[w], v);
      }
    }
  }
}

int grid_for_l(long work) {
  long g = (work + kBlock - 1) / kBlock;
  long cap = 256L * 8;
  if (g > cap) g = cap;
  if (g < 1) g = 1;
  return (int)g;
}

void check_lf(const at::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
  TORCH_CHECK(t.scalar_type() == at::kFloat || t.scalar_type() == at::kDouble,
              name, " must be float32/float64");
}

}  // namespace

at::Tensor lift_head_fwd(const at::Tensor& x, const at::Tensor& W1,
                         const at::Tensor& b1, const at::Tensor& W2,
                         const at::Tensor& b2) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous(), "x must be contiguous GPU");
  if (x.scalar_type() != at::kBFloat16) { check_lf(x, "x"); }
  TORCH_CHECK(x.dim() == 3, "x must be [B,C,S]");
  int B = (int)x.size(0), C = (int)x.size(1);
  long S = x.size(2);
  int Tn = (int)W1.size(0), W = (int)W2.size(0);
  TORCH_CHECK(W1.size(1) == 1 && (int)W2.size(1) == C, "lift_head shapes");
  TORCH_CHECK(C <= 4 && Tn <= 32 && W <= 24, "lift_head: unsupported dims");

  auto out = at::empty({B, W, S, (long)Tn}, x.options());
  if (x.numel() == 0) return out;
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  int grid = grid_for_l((long)B * S);
  if (x.scalar_type() == at::kFloat || x.scalar_type() == at::kBFloat16) {
    int grid2 = grid_for_l(2 * (((long)B * S + 1) / 2) * 32);
    const bool bf16 = x.scalar_type() == at::kBFloat16;
    auto W1f = bf16 ? W1.to(at::kFloat).contiguous() : W1;
    auto b1f = bf16 ? b1.to(at::kFloat).contiguous() : b1;
    auto W2f = bf16 ? W2.to(at::kFloat).contiguous() : W2;
    auto b2f = bf16 ? b2.to(at::kFloat).contiguous() : b2;
#define LH_FWD(WT, TIO)                                                       \
    hipLaunchKernelGGL((lift_head_fwd_lk_kernel<WT, TIO>), dim3(grid2),       \
                       dim3(kBlock), 0, stream,                               \
                       reinterpret_cast<const TIO*>(x.data_ptr()),            \
                       W1f.data_ptr<float>(), b1f.data_ptr<float>(),          \
                       W2f.data_ptr<float>(), b2f.data_ptr<float>(),          \
                       reinterpret_cast<TIO*>(out.data_ptr()), B, C, W, Tn, S)
    if (bf16) { if (W == 20) LH_FWD(20, unsigned short);
                else LH_FWD(0, unsigned short); }
    else { if (W == 20) LH_FWD(20, float); else LH_FWD(0, float); }
#undef LH_FWD
    DFNO_CHECK_LAUNCH("lift_head");
    return out;
  }
  AT_DISPATCH_FLOATING_TYPES(x.scalar_type(), "lift_head_fwd", [&] {
    if (W == 20) {
      hipLaunchKernelGGL((lift_head_fwd_kernel<scalar_t, 4, 32, 24, 20>), dim3(grid),
                         dim3(kBlock), 0, stream, x.data_ptr<scalar_t>(),
                         W1.data_ptr<scalar_t>(), b1.data_ptr<scalar_t>(),
                         W2.data_ptr<scalar_t>(), b2.data_ptr<scalar_t>(),
                         out.data_ptr<scalar_t>(), B, C, W, Tn, S);
    } else {
      hipLaunchKernelGGL((lift_head_fwd_kernel<scalar_t, 4, 32, 24>), dim3(grid),
                         dim3(kBlock), 0, stream, x.data_ptr<scalar_t>(),
                         W1.data_ptr<scalar_t>(), b1.data_ptr<scalar_t>(),
                         W2.data_ptr<scalar_t>(), b2.data_ptr<scalar_t>(),
                         out.data_ptr<scalar_t>(), B, C, W, Tn, S);
    }
  });
  DFNO_CHECK_LAUNCH("lift_head");
  return out;
}

std::vector<at::Tensor> lift_head_bwd(const at::Tensor& gy, const at::Tensor& x,
                                      const at::Tensor& W1, const at::Tensor& b1,
                                      const at::Tensor& W2, const at::Tensor& b2) {
  TORCH_CHECK(gy.is_cuda() && gy.is_contiguous() && x.is_contiguous() &&
              gy.scalar_type() == x.scalar_type(), "lift_head_bwd IO");
  if (x.scalar_type() != at::kBFloat16) { check_lf(gy, "gy"); check_lf(x, "x"); }
  int B = (int)x.size(0), C = (int)x.size(1);
  long S = x.size(2);
  int Tn = (int)W1.size(0), W = (int)W2.size(0);
  TORCH_CHECK(C <= 4 && Tn <= 32 && W <= 24, "lift_head: unsupported dims");

  auto gx = at::empty_like(x);
  // weight grads accumulate fp32 even for bf16 activations
  auto fopt = x.options().dtype(x.scalar_type() == at::kDouble
                                    ? at::kDouble : at::kFloat);
  auto gW1 = at::zeros({Tn, 1}, fopt);
  auto gb1 = at::zeros({Tn}, fopt);
  auto gW2 = at::zeros({W, C}, fopt);
  auto gb2 = at::zeros({W}, fopt);
  if (x.numel() == 0) return {gx, gW1, gb1, gW2, gb2};

  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  int grid = grid_for_l((long)B * S);
  if (x.scalar_type() == at::kFloat || x.scalar_type() == at::kBFloat16) {
    int grid2 = grid_for_l(2 * (((long)B * S + 1) / 2) * 32);
    const bool bf16 = x.scalar_type() == at::kBFloat16;
    auto W1f = bf16 ? W1.to(at::kFloat).contiguous() : W1;
    auto b1f = bf16 ? b1.to(at::kFloat).contiguous() : b1;
    auto W2f = bf16 ? W2.to(at::kFloat).contiguous() : W2;
    auto b2f = bf16 ? b2.to(at::kFloat).contiguous() : b2;
#define LH_BWD(WT, TIO)                                                       \
    hipLaunchKernelGGL((lift_head_bwd_lk_kernel<WT, TIO>), dim3(grid2),       \
                       dim3(kBlock), 0, stream,                               \
                       reinterpret_cast<const TIO*>(gy.data_ptr()),           \
                       reinterpret_cast<const TIO*>(x.data_ptr()),            \
                       W1f.data_ptr<float>(), b1f.data_ptr<float>(),          \
                       W2f.data_ptr<float>(), b2f.data_ptr<float>(),          \
                       reinterpret_cast<TIO*>(gx.data_ptr()),                 \
                       gW1.data_ptr<float>(), gb1.data_ptr<float>(),          \
                       gW2.data_ptr<float>(), gb2.data_ptr<float>(),          \
                       B, C, W, Tn, S)
    if (bf16) { if (W == 20) LH_BWD(20, unsigned short);
                else LH_BWD(0, unsigned short); }
    else { if (W == 20) LH_BWD(20, float); else LH_BWD(0, float); }
#undef LH_BWD
    DFNO_CHECK_LAUNCH("lift_head");
    return {gx, gW1, gb1, gW2, gb2};
  }
  AT_DISPATCH_FLOATING_TYPES(x.scalar_type(), "lift_head_bwd", [&] {
    if (W == 20) {
      hipLaunchKernelGGL((lift_head_bwd_kernel<scalar_t, 4, 32, 24, 20>), dim3(grid),
                         dim3(kBlock), 0, stream, gy.data_ptr<scalar_t>(),
                         x.data_ptr<scalar_t>(), W1.data_ptr<scalar_t>(),
                         b1.data_ptr<scalar_t>(), W2.data_ptr<scalar_t>(),
                         b2.data_ptr<scalar_t>(), gx.data_ptr<scalar_t>(),
                         gW1.data_ptr<scalar_t>(), gb1.data_ptr<scalar_t>(),
                         gW2.data_ptr<scalar_t>(), gb2.data_ptr<scalar_t>(),
                         B, C, W, Tn, S);
    } else {
      hipLaunchKernelGGL((lift_head_bwd_kernel<scalar_t, 4, 32, 24>), dim3(grid),
                         dim3(kBlock), 0, stream, gy.data_ptr<scalar_t>(),
                         x.data_ptr<scalar_t>(), W1.data_ptr<scalar_t>(),
                         b1.data_ptr<scalar_t>(), W2.data_ptr<scalar_t>(),
                         b2.data_ptr<scalar_t>(), gx.data_ptr<scalar_t>(),
                         gW1.data_ptr<scalar_t>(), gb1.data_ptr<scalar_t>(),
                         gW2.data_ptr<scalar_t>(), gb2.data_ptr<scalar_t>(),
                         B, C, W, Tn, S);
    }
  });
  DFNO_CHECK_LAUNCH("lift_head");
  return {gx, gW1, gb1, gW2, gb2};
}
