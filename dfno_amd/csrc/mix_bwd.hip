// Fused backward for the trunk 20x20 channel mixes (gfx950, fp32).
//
// The unfused chain per mix-backward is three full-activation kernels:
// gelu_bwd (read gy+z, write gz), channel_mix_fwd_t (read gz, write gx)
// and gw_outer (read gz+x) — ~4.4 GB of HBM traffic at the flagship
// (z/gy/x are [1,20,64^3*30] fp32).  Here gz exists only as a [20 x 64]
// LDS tile computed while staging gy/z; gx = W^T @ gz and
// gW = gz @ x^T (+ gb via a constant ones column) run as
// v_mfma_f32_16x16x4 over the tile, with the gW fragments carried in
// registers across tiles and flushed once per block (atomicAdd, the
// same fp32 reduction-order caveat as every gw kernel here).  When the
// gz tensor itself is a needed gradient (linear_res_gelu's residual
// input), WG=true streams the tile back coalesced — still one pass.
//
// Reference semantics: dfno.py:348-352 (the per-block linear+GELU pair
// around the spectral conv).

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <torch/extension.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>

#include "kernels.h"
#include "gelu_math.h"

namespace {

constexpr int kBlock = 256;

typedef float f32x4_mb __attribute__((ext_vector_type(4)));

__device__ __forceinline__ float mb_ld(const float* p) { return *p; }
__device__ __forceinline__ float mb_ld(const unsigned short* p) {
  return __uint_as_float(((unsigned int)*p) << 16);
}
__device__ __forceinline__ void mb_st(float* p, float v) { *p = v; }
__device__ __forceinline__ void mb_st(unsigned short* p, float v) {
  __hip_bfloat16 h = __float2bfloat16(v);
  *p = *reinterpret_cast<unsigned short*>(&h);
}

// TIO = unsigned short selects bf16 activations (fp32 compute; W and the
// gW/gb accumulators stay fp32).  PH: phase mask for perf bisection
// (1 staging+gz, +2 gx, +4 gW); production uses 7.
template <typename TIO, bool WG, int PH = 7, int TS = 64>
__global__ __launch_bounds__(kBlock, 4) void mix_bwd_fused_kernel(
    const TIO* __restrict__ gy, const TIO* __restrict__ z,
    const TIO* __restrict__ x, const float* __restrict__ W,
    TIO* __restrict__ gx, float* __restrict__ gW, float* __restrict__ gb,
    TIO* __restrict__ gzout, int B, long S) {
  constexpr int C = 20;            // in = out channels (trunk width)
  constexpr int LD = TS + 4;
  extern __shared__ __align__(16) char smem_raw[];
  float* gzt = reinterpret_cast<float*>(smem_raw);   // [C][LD]
  float* xt = gzt + (size_t)C * LD;                  // [C][LD]

  const int lane = (int)(threadIdx.x & 63);
  const int wave = (int)(threadIdx.x >> 6);
  const int l16 = lane & 15;
  const int kg = lane >> 4;

  // gW/gb fragments: M = 2 o-tiles, N = 2 tiles (i channels + ones col)
  f32x4_mb wacc;
  wacc = f32x4_mb{0.f, 0.f, 0.f, 0.f};

  const long stiles = (S + TS - 1) / TS;
  const long tend = (long)B * stiles;
  // register-prefetched staging (proj_head pattern): next tile's gy/z/x
  // fly over this tile's MFMA phases
  constexpr int NPF = (2 * C * TS + kBlock - 1) / kBlock;
  float pfa[NPF], pfb[NPF];        // pfa: x or gy row; pfb: z row
  auto prefetch = [&](long tt) {
    if (tt >= tend) return;
    const int b = (int)(tt / stiles);
    const long s0 = (tt % stiles) * TS;
    const int nv = (int)min((long)TS, S - s0);
#pragma unroll
    for (int q = 0; q < NPF; ++q) {
      const int r = (int)threadIdx.x + q * kBlock;
      const int row = r / TS;
      const int c = r - row * TS;
      float a = 0.f, bb = 0.f;
      if (c < nv) {
        if (row < C) {
          a = mb_ld(x + ((long)b * C + row) * S + s0 + c);
        } else {
          const long off = ((long)b * C + (row - C)) * S + s0 + c;
          a = mb_ld(gy + off);
          bb = mb_ld(z + off);
        }
      }
      pfa[q] = a;
      pfb[q] = bb;
    }
  };
  prefetch(blockIdx.x);

  for (long t = blockIdx.x; t < tend; t += gridDim.x) {
    const int b = (int)(t / stiles);
    const long s0 = (t % stiles) * TS;
    const int nv = (int)min((long)TS, S - s0);
    __syncthreads();               // prior tile's phase reads done
#pragma unroll
    for (int q = 0; q < NPF; ++q) {
      const int r = (int)threadIdx.x + q * kBlock;
      const int row = r / TS;
      const int c = r - row * TS;
      if (row < C) {
        xt[row * LD + c] = pfa[q];
      } else {
        const float gzv = pfa[q] * dfno_gelu::gelu_grad(pfb[q]);
        gzt[(row - C) * LD + c] = gzv;
        if (WG && c < nv)
          mb_st(gzout + ((long)b * C + (row - C)) * S + s0 + c, gzv);
      }
    }
    prefetch(t + gridDim.x);
    __syncthreads();
    if constexpr ((PH & 2) == 0 && (PH & 4) == 0) continue;
    // phase A: gx = W^T @ gz.  A[m=i][k=o] = W[o*C+i] (per-lane global,
    // L1-resident), B[n=c][k=o] = gzt[o][c]; K = C padded to 24.
#pragma unroll
    for (int pp = 0; pp < ((PH & 2) ? (2 * (TS / 16)) / 4 : 0); ++pp) {
      const int p = wave + 4 * pp;
      const int mt = p / (TS / 16), nt = p % (TS / 16);
      const int m = mt * 16 + l16;
      const int n = nt * 16 + l16;
      const bool av = m < C;
      f32x4_mb c4 = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int k = 0; k < C; k += 4) {       // C = 20: K is 4-aligned
        const int ko = k + kg;
        const float a = av ? W[ko * C + m] : 0.f;
        const float bb = gzt[ko * LD + n];
        c4 = __builtin_amdgcn_mfma_f32_16x16x4f32(a, bb, c4, 0, 0, 0);
      }
      const long sc = s0 + nt * 16 + l16;
      if (sc < S) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int i = mt * 16 + kg * 4 + r;
          if (i < C) mb_st(gx + ((long)b * C + i) * S + sc, c4[r]);
        }
      }
    }
    // phase B: gW fragments += gz @ [x; ones]^T.  One (mt, nt) pair per
    // wave: mt = wave>>1 (o-tiles), nt = wave&1 (i-tiles; col C = ones
    // for gb).
    if constexpr ((PH & 4) != 0) {
      const int mt = wave >> 1, nt = wave & 1;
      const float* gr = gzt + (mt * 16 + l16) * LD;
      const int ncol = nt * 16 + l16;
      const float* xr = xt + ncol * LD;
      const bool av = (mt * 16 + l16) < C;
      const bool bx = ncol < C;
      const bool bones = ncol == C;
#pragma unroll 8
      for (int k = 0; k < TS; k += 4) {
        const float a = av ? gr[k + kg] : 0.f;
        const float bb = bx ? xr[k + kg] : (bones ? 1.f : 0.f);
        wacc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, bb, wacc, 0, 0, 0);
      }
    }
  }

  // flush gW/gb fragments.  D[m = o][n]: n < C -> gW[o][n], n == C -> gb[o]
  {
    const int mt = wave >> 1, nt = wave & 1;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int o = mt * 16 + kg * 4 + r;
      const int n = nt * 16 + l16;
      const float v = wacc[r];
      if (o < C && v != 0.f) {
        if (n < C) atomicAdd(&gW[(size_t)o * C + n], v);
        else if (n == C && gb != nullptr) atomicAdd(&gb[o], v);
      }
    }
  }
}

}  // namespace

std::vector<at::Tensor> channel_mix_bwd_fused(const at::Tensor& gy,
                                              const at::Tensor& z,
                                              const at::Tensor& x,
                                              const at::Tensor& W,
                                              bool want_bias, bool want_gz) {
  const bool bf16 = gy.scalar_type() == at::kBFloat16;
  TORCH_CHECK(gy.is_cuda() && gy.is_contiguous() && z.is_contiguous() &&
              x.is_contiguous() && W.is_contiguous() &&
              (bf16 || gy.scalar_type() == at::kFloat) &&
              x.scalar_type() == gy.scalar_type() &&
              z.scalar_type() == gy.scalar_type() &&
              W.scalar_type() == at::kFloat,
              "mix_bwd_fused: fp32/bf16 activations, fp32 W");
  int B = (int)x.size(0), I = (int)x.size(1);
  long S = x.size(2);
  int O = (int)W.size(0);
  TORCH_CHECK(I == 20 && O == 20, "mix_bwd_fused: trunk 20x20 only");
  TORCH_CHECK(gy.numel() == x.numel() && z.numel() == x.numel(),
              "mix_bwd_fused: shape mismatch");

  auto gx = at::empty({B, I, S}, x.options());
  auto fopt = x.options().dtype(at::kFloat);   // gW/gb accumulate fp32
  auto gW = at::zeros({O, I}, fopt);
  auto gb = want_bias ? at::zeros({O}, fopt) : at::empty({0}, fopt);
  auto gz = want_gz ? at::empty({B, O, S}, x.options())
                    : at::empty({0}, x.options());
  if (x.numel() == 0) return {gx, gW, gb, gz};

  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  static const bool ts128 = []() {
    const char* e = getenv("DFNO_MIX_TS128");    // tile-size A/B knob
    return e && e[0] == '1';
  }();
  const int TS = ts128 ? 128 : 64;
  size_t smem = sizeof(float) * 2 * 20 * (TS + 4);
  long stiles = (S + TS - 1) / TS;
  int grid = (int)std::min((long)B * stiles, 1024L);
  float* gbp = want_bias ? gb.data_ptr<float>() : nullptr;
  static const int ph = []() {
    const char* e = getenv("DFNO_MIX_PHASES");   // perf-bisect knob
    return e ? atoi(e) : 7;
  }();
#define MB_LAUNCH(TIO, WG, GZP) MB_LAUNCH_T(TIO, WG, GZP, 7, 64)
#define MB_LAUNCH_P(TIO, WG, GZP, PHV) MB_LAUNCH_T(TIO, WG, GZP, PHV, 64)
#define MB_LAUNCH_T(TIO, WG, GZP, PHV, TSV)                                  \
  hipLaunchKernelGGL((mix_bwd_fused_kernel<TIO, WG, PHV, TSV>), dim3(grid), \
                     dim3(kBlock), smem, stream,                             \
                     reinterpret_cast<const TIO*>(gy.data_ptr()),            \
                     reinterpret_cast<const TIO*>(z.data_ptr()),             \
                     reinterpret_cast<const TIO*>(x.data_ptr()),             \
                     W.data_ptr<float>(),                                    \
                     reinterpret_cast<TIO*>(gx.data_ptr()),                  \
                     gW.data_ptr<float>(), gbp, GZP, B, S)
  if (bf16) {
    auto gzp = want_gz ? reinterpret_cast<unsigned short*>(gz.data_ptr())
                       : nullptr;
    if (want_gz) MB_LAUNCH(unsigned short, true, gzp);
    else         MB_LAUNCH(unsigned short, false, nullptr);
  } else {
    auto gzp = want_gz ? reinterpret_cast<float*>(gz.data_ptr()) : nullptr;
    if (want_gz) {
      if (ts128) { MB_LAUNCH_T(float, true, gzp, 7, 128); }
      else if (ph == 1) { MB_LAUNCH_T(float, true, gzp, 1, 64); }
      else if (ph == 3) { MB_LAUNCH_T(float, true, gzp, 3, 64); }
      else if (ph == 5) { MB_LAUNCH_T(float, true, gzp, 5, 64); }
      else { MB_LAUNCH_T(float, true, gzp, 7, 64); }
    }
    else         MB_LAUNCH(float, false, nullptr);
  }
#undef MB_LAUNCH_T
#undef MB_LAUNCH_P
#undef MB_LAUNCH
  DFNO_CHECK_LAUNCH("mix_bwd_fused");
  return {gx, gW, gb, gz};
}
