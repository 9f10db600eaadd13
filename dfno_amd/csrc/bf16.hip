// gfx950 bf16-storage pointwise kernels (the bf16 compute config,
// BASELINE.json config #2 / VERDICT.md round-1 item 3).
//
// Storage is bf16 (halves every activation's HBM traffic vs the fp32
// headline config — these ops are bandwidth-bound at 8 TB/s HBM3E);
// arithmetic is fp32 (convert-on-load, round-to-nearest-even on store).
// Vector IO is 8 bf16 per lane per instruction (uint4 = 16 B, the
// coalescing sweet spot; cdna_hip_programming.md Guideline 13: hipcc does
// NOT auto-vectorize bf16 loads).
//
// Kernels:
//  * bf16_channel_mix (x-resident, I <= 32): y[b,o,s] = act(sum_i W[o,i]
//    x[b,i,s] + bias[o] [+ res]) — covers the width-20 channel mixes, the
//    20->128 projection lift and (transposed) grad-x.
//  * bf16_channel_mix_ores (O <= 8, streams I): the 128->1 projection.
//  * bf16_gelu fwd/bwd, bf16_add_gelu elementwise epilogues.
//  * bf16_gw: split-s grad-W/grad-bias outer-product reduction with fp32
//    accumulation (atomics on an fp32 staging buffer).

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <torch/extension.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>

#include "kernels.h"
#include "gelu_math.h"

namespace {

constexpr int kBlock = 256;
constexpr int kVec = 8;   // bf16 per lane per vector op (16 B)

__device__ __forceinline__ float b2f(unsigned short h) {
  unsigned int u = ((unsigned int)h) << 16;
  return __uint_as_float(u);
}

// Pack a float pair to bf16x2 with round-to-nearest-even: the compiler
// lowers the __hip_bfloat162 conversion to v_cvt_pk_bf16_f32 (one VALU op
// per pair; the first integer-math version cost ~5 VALU per element and
// made every store-side kernel conversion-bound).
__device__ __forceinline__ unsigned int f2b2(float lo, float hi) {
  __hip_bfloat162 h2 = __float22bfloat162_rn(float2{lo, hi});
  return *reinterpret_cast<unsigned int*>(&h2);
}

__device__ __forceinline__ void load8(const unsigned short* p, float* out) {
  const uint4 raw = *reinterpret_cast<const uint4*>(p);
  const unsigned int w[4] = {raw.x, raw.y, raw.z, raw.w};
#pragma unroll
  for (int k = 0; k < 4; ++k) {
    out[2 * k] = b2f((unsigned short)(w[k] & 0xffffu));
    out[2 * k + 1] = b2f((unsigned short)(w[k] >> 16));
  }
}

__device__ __forceinline__ void store8(unsigned short* p, const float* v) {
  uint4 raw;
  raw.x = f2b2(v[0], v[1]);
  raw.y = f2b2(v[2], v[3]);
  raw.z = f2b2(v[4], v[5]);
  raw.w = f2b2(v[6], v[7]);
  *reinterpret_cast<uint4*>(p) = raw;
}

__device__ __forceinline__ void load4(const unsigned short* p, float* out) {
  const uint2 raw = *reinterpret_cast<const uint2*>(p);
  const unsigned int w[2] = {raw.x, raw.y};
#pragma unroll
  for (int k = 0; k < 2; ++k) {
    out[2 * k] = b2f((unsigned short)(w[k] & 0xffffu));
    out[2 * k + 1] = b2f((unsigned short)(w[k] >> 16));
  }
}

__device__ __forceinline__ void store4(unsigned short* p, const float* v) {
  uint2 raw;
  raw.x = f2b2(v[0], v[1]);
  raw.y = f2b2(v[2], v[3]);
  *reinterpret_cast<uint2*>(p) = raw;
}

// ---------------------------------------------------------------------------
// x-resident channel mix (I <= IMAX): one thread owns kVec consecutive s.
// ---------------------------------------------------------------------------

// IT/OT > 0 pin the channel counts at compile time (runtime trip counts
// serialize on uniform-load waits — the round-1 folding lesson; the fp32
// twins carry the same pins)
template <int IMAX, int VEC, bool ACT, int IT = 0, int OT = 0>
__global__ __launch_bounds__(kBlock) void bf16_channel_mix_xres_kernel(
    const unsigned short* __restrict__ x, const unsigned short* __restrict__ W,
    const unsigned short* __restrict__ bias, unsigned short* __restrict__ y,
    unsigned short* __restrict__ z, int B, int I_, int O_, long S, bool wt,
    bool has_bias, bool write_z, const unsigned short* __restrict__ res) {
  const int I = IT > 0 ? IT : I_;
  const int O = OT > 0 ? OT : O_;
  // VEC=4 for the wide-IMAX variants (xr[24][8] = 192 VGPRs would spill)
  extern __shared__ __align__(16) char smem_raw[];
  float* Wl = reinterpret_cast<float*>(smem_raw);   // [O*I] fp32
  float* bl = Wl + (size_t)O * I;                   // [O]
  for (int k = threadIdx.x; k < O * I; k += blockDim.x) Wl[k] = b2f(W[k]);
  if (has_bias)
    for (int k = threadIdx.x; k < O; k += blockDim.x) bl[k] = b2f(bias[k]);
  __syncthreads();

  auto loadv = [](const unsigned short* p, float* v) {
    if constexpr (VEC == 8) load8(p, v); else load4(p, v);
  };
  auto storev = [](unsigned short* p, const float* v) {
    if constexpr (VEC == 8) store8(p, v); else store4(p, v);
  };

  long nchunks = S / VEC;   // host guarantees S % 8 == 0
  long t0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;

  for (long t = t0; t < (long)B * nchunks; t += stride) {
    int b = (int)(t / nchunks);
    long s = (t % nchunks) * VEC;

    float xr[IMAX][VEC];
    const unsigned short* xb = x + ((long)b * I) * S + s;
#pragma unroll
    for (int i = 0; i < IMAX; ++i) {
      if (i < I) loadv(xb + (long)i * S, xr[i]);
    }

    unsigned short* yb = y + ((long)b * O) * S + s;
    unsigned short* zb = write_z ? z + ((long)b * O) * S + s : nullptr;
#pragma unroll 4
    for (int o = 0; o < (OT > 0 ? OT : 512); ++o) {
      if (OT == 0 && o >= O) break;
      float acc[VEC];
      const float bv = has_bias ? bl[o] : 0.f;
#pragma unroll
      for (int k = 0; k < VEC; ++k) acc[k] = bv;
#pragma unroll
      for (int i = 0; i < IMAX; ++i) {
        if (i < I) {
          const float wv = wt ? Wl[(size_t)i * O + o] : Wl[(size_t)o * I + i];
#pragma unroll
          for (int k = 0; k < VEC; ++k) acc[k] += wv * xr[i][k];
        }
      }
      if (res != nullptr) {
        float rv[VEC];
        loadv(res + ((long)b * O + o) * S + s, rv);
#pragma unroll
        for (int k = 0; k < VEC; ++k) acc[k] += rv[k];
      }
      if (write_z) storev(zb + (long)o * S, acc);
      if (ACT) {
#pragma unroll
        for (int k = 0; k < VEC; ++k) acc[k] = dfno_gelu::gelu(acc[k]);
      }
      storev(yb + (long)o * S, acc);
    }
  }
}

// ---------------------------------------------------------------------------
// accumulator-resident channel mix (O <= OMAX, streams I) — the 128->1 head
// ---------------------------------------------------------------------------

template <int OMAX, int VEC, bool ACT, int IT = 0, int OT = 0>
__global__ __launch_bounds__(kBlock) void bf16_channel_mix_ores_kernel(
    const unsigned short* __restrict__ x, const unsigned short* __restrict__ W,
    const unsigned short* __restrict__ bias, unsigned short* __restrict__ y,
    unsigned short* __restrict__ z, int B, int I_, int O_, long S, bool wt,
    bool has_bias, bool write_z, const unsigned short* __restrict__ res) {
  const int I = IT > 0 ? IT : I_;
  const int O = OT > 0 ? OT : O_;
  // VEC=4 for the wide-OMAX variants: acc[24][8] would eat 192 VGPRs and
  // spill; [24][4] keeps the whole accumulator set resident.
  extern __shared__ __align__(16) char smem_raw[];
  float* Wl = reinterpret_cast<float*>(smem_raw);   // [O*I]
  float* bl = Wl + (size_t)O * I;
  for (int k = threadIdx.x; k < O * I; k += blockDim.x) Wl[k] = b2f(W[k]);
  if (has_bias)
    for (int k = threadIdx.x; k < O; k += blockDim.x) bl[k] = b2f(bias[k]);
  __syncthreads();

  auto loadv = [](const unsigned short* p, float* v) {
    if constexpr (VEC == 8) load8(p, v); else load4(p, v);
  };
  auto storev = [](unsigned short* p, const float* v) {
    if constexpr (VEC == 8) store8(p, v); else store4(p, v);
  };

  long nchunks = S / VEC;
  long t0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;

  for (long t = t0; t < (long)B * nchunks; t += stride) {
    int b = (int)(t / nchunks);
    long s = (t % nchunks) * VEC;

    float acc[OMAX][VEC];
#pragma unroll
    for (int o = 0; o < OMAX; ++o) {
      const float bv = (o < O && has_bias) ? bl[o] : 0.f;
#pragma unroll
      for (int k = 0; k < VEC; ++k) acc[o][k] = bv;
    }

    const unsigned short* xb = x + ((long)b * I) * S + s;
#pragma unroll 4
    for (int i = 0; i < I; ++i) {
      float xv[VEC];
      loadv(xb + (long)i * S, xv);
#pragma unroll
      for (int o = 0; o < OMAX; ++o) {
        if (o < O) {
          const float wv = wt ? Wl[(size_t)i * O + o] : Wl[(size_t)o * I + i];
#pragma unroll
          for (int k = 0; k < VEC; ++k) acc[o][k] += wv * xv[k];
        }
      }
    }

    unsigned short* yb = y + ((long)b * O) * S + s;
    unsigned short* zb = write_z ? z + ((long)b * O) * S + s : nullptr;
#pragma unroll
    for (int o = 0; o < OMAX; ++o) {
      if (o < O) {
        if (res != nullptr) {
          float rv[VEC];
          loadv(res + ((long)b * O + o) * S + s, rv);
#pragma unroll
          for (int k = 0; k < VEC; ++k) acc[o][k] += rv[k];
        }
        if (write_z) storev(zb + (long)o * S, acc[o]);
        if (ACT) {
#pragma unroll
          for (int k = 0; k < VEC; ++k) acc[o][k] = dfno_gelu::gelu(acc[o][k]);
        }
        storev(yb + (long)o * S, acc[o]);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// elementwise gelu / add-gelu (8-wide)
// ---------------------------------------------------------------------------

__global__ void bf16_gelu_fwd_kernel(const unsigned short* __restrict__ x,
                                     unsigned short* __restrict__ y, long n8) {
  long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = i0; i < n8; i += stride) {
    float v[kVec];
    load8(x + i * kVec, v);
#pragma unroll
    for (int k = 0; k < kVec; ++k) v[k] = dfno_gelu::gelu(v[k]);
    store8(y + i * kVec, v);
  }
}

__global__ void bf16_gelu_bwd_kernel(const unsigned short* __restrict__ gy,
                                     const unsigned short* __restrict__ z,
                                     unsigned short* __restrict__ gz, long n8) {
  long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = i0; i < n8; i += stride) {
    float g[kVec], zv[kVec];
    load8(gy + i * kVec, g);
    load8(z + i * kVec, zv);
#pragma unroll
    for (int k = 0; k < kVec; ++k) g[k] *= dfno_gelu::gelu_grad(zv[k]);
    store8(gz + i * kVec, g);
  }
}

__global__ void bf16_add_gelu_kernel(const unsigned short* __restrict__ a,
                                     const unsigned short* __restrict__ b,
                                     unsigned short* __restrict__ y,
                                     unsigned short* __restrict__ z, long n8) {
  long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = i0; i < n8; i += stride) {
    float av[kVec], bv[kVec];
    load8(a + i * kVec, av);
    load8(b + i * kVec, bv);
#pragma unroll
    for (int k = 0; k < kVec; ++k) av[k] += bv[k];
    store8(z + i * kVec, av);
#pragma unroll
    for (int k = 0; k < kVec; ++k) av[k] = dfno_gelu::gelu(av[k]);
    store8(y + i * kVec, av);
  }
}

// ---------------------------------------------------------------------------
// grad-W (+grad-bias): gW[o,i] = sum_{b,s} gz[b,o,s] x[b,i,s], fp32 accum.
// LDS-tiled outer product: each s-tile of TS elements is staged (converted
// to fp32, row stride TS+1 so lanes with consecutive i hit distinct banks),
// then every thread owns ceil(O*I/256) fixed (o,i) pairs and accumulates in
// REGISTERS over the tile — no atomics in the hot loop (the first version's
// per-element LDS atomics serialized to 14 ms/launch); one global atomicAdd
// per pair at thread end.
// ---------------------------------------------------------------------------

template <int NP>
__global__ __launch_bounds__(kBlock) void bf16_gw_kernel(
    const unsigned short* __restrict__ gz, const unsigned short* __restrict__ x,
    float* __restrict__ gW, float* __restrict__ gb,
    int B, int I, int O, long S, bool want_bias) {
  constexpr int TS = 128;          // s-elements per tile
  constexpr int LD = TS + 4;       // row stride: float4 reads stay 16 B-
                                   // aligned, banks spread (132 % 64 = 4)
  constexpr int OSL = 32;          // O-slab per blockIdx.y: caps the LDS
                                   // footprint at (32+32)*132*4 = 33 KiB so
                                   // 4 blocks/CU stay resident (the O=128
                                   // single-slab version ran 1 block/CU)
  extern __shared__ __align__(16) char smem_raw[];
  float* xs = reinterpret_cast<float*>(smem_raw);    // [I][LD]
  float* gs = xs + (size_t)32 * LD;                  // [O_sl][LD]

  const int o0 = (int)blockIdx.y * OSL;
  const int O_sl = min(OSL, O - o0);

  // fixed (o, i) pairs of this thread within the slab
  int po[NP], pi[NP];
  float acc[NP];
#pragma unroll
  for (int p = 0; p < NP; ++p) {
    const int idx = (int)threadIdx.x + p * kBlock;
    po[p] = idx / I;
    pi[p] = idx - po[p] * I;
    acc[p] = 0.f;
  }
  float accb = 0.f;                // bias partial (thread o = threadIdx.x)

  const long ntiles = (S / TS) * (long)B;   // host guarantees S % TS == 0
  for (long t = blockIdx.x; t < ntiles; t += gridDim.x) {
    const int b = (int)(t / (S / TS));
    const long s0 = (t % (S / TS)) * TS;
    // stage: 8 bf16 per thread per instruction, convert to fp32
    for (int r = threadIdx.x; r < (I + O_sl) * (TS / kVec); r += kBlock) {
      const int row = r / (TS / kVec);
      const int col = (r - row * (TS / kVec)) * kVec;
      float v[kVec];
      if (row < I) {
        load8(x + ((long)b * I + row) * S + s0 + col, v);
#pragma unroll
        for (int k = 0; k < kVec; ++k) xs[row * LD + col + k] = v[k];
      } else {
        const int ro = row - I;
        load8(gz + ((long)b * O + o0 + ro) * S + s0 + col, v);
#pragma unroll
        for (int k = 0; k < kVec; ++k) gs[ro * LD + col + k] = v[k];
      }
    }
    __syncthreads();
#pragma unroll
    for (int p = 0; p < NP; ++p) {
      if (po[p] < O_sl) {
        const float* gr = gs + po[p] * LD;
        const float* xr = xs + pi[p] * LD;
#pragma unroll
        for (int k = 0; k < TS; k += 4) {
          const float4 g4 = *reinterpret_cast<const float4*>(gr + k);
          const float4 x4 = *reinterpret_cast<const float4*>(xr + k);
          acc[p] += g4.x * x4.x + g4.y * x4.y + g4.z * x4.z + g4.w * x4.w;
        }
      }
    }
    if (want_bias && (int)threadIdx.x < O_sl) {
      const float* gr = gs + threadIdx.x * LD;
#pragma unroll
      for (int k = 0; k < TS; k += 4) {
        const float4 g4 = *reinterpret_cast<const float4*>(gr + k);
        accb += g4.x + g4.y + g4.z + g4.w;
      }
    }
    __syncthreads();
  }
#pragma unroll
  for (int p = 0; p < NP; ++p)
    if (po[p] < O_sl) atomicAdd(&gW[(size_t)(o0 + po[p]) * I + pi[p]], acc[p]);
  if (want_bias && (int)threadIdx.x < O_sl)
    atomicAdd(&gb[o0 + threadIdx.x], accb);
}

// ---------------------------------------------------------------------------
// MFMA grad-W: gW[o,i] = sum_s gz[o,s] x[i,s] on v_mfma_f32_16x16x32_bf16.
// The one truly GEMM-shaped hot op of the model (K = S ~ 10^7): the VALU
// version above is arithmetic-limited (~10 flop per bf16-byte at width 20,
// right at the fp32 VALU roofline); the matrix cores run the same contraction
// at the 2.5 PF bf16 rate, making it memory-bound.  Fragment loads need no
// LDS: for both A (gz) and B (x) a lane's 8 k-elements are 8 CONSECUTIVE
// s-positions of one channel row — one 16-byte load each, rows spread over
// lanes (lane groups at k-offsets 0/8/16/24 tile 64 B per row).  Bias falls
// out of one extra MFMA against a ones-column B fragment.
// ---------------------------------------------------------------------------

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

__device__ __forceinline__ bf16x8 load_frag(const unsigned short* row_base,
                                            long k0) {
  // 8 consecutive bf16 at s = k0 (16 B)
  return *reinterpret_cast<const bf16x8*>(row_base + k0);
}

template <int NPAIR>
__global__ __launch_bounds__(kBlock) void bf16_gw_mfma_kernel(
    const unsigned short* __restrict__ gz, const unsigned short* __restrict__ x,
    float* __restrict__ gW, float* __restrict__ gb,
    int B, int I, int O, long S, bool want_bias) {
  constexpr int TS = 256;          // s per grid tile (8 MFMA steps of K=32)
  const int lane = (int)(threadIdx.x & 63);
  const int wave = (int)(threadIdx.x >> 6);
  const int row16 = lane & 15;     // A row / B col within the 16-tile
  const int kgrp = lane >> 4;      // k-group (8 k each)

  const int oT = (O + 15) / 16, iT = (I + 15) / 16;
  const int npairs = oT * iT;      // host guarantees npairs <= 4 * NPAIR

  f32x4 acc[NPAIR];
  f32x4 accb[NPAIR];               // bias (only i-tile 0 pairs contribute)
#pragma unroll
  for (int p = 0; p < NPAIR; ++p) {
    acc[p] = f32x4{0.f, 0.f, 0.f, 0.f};
    accb[p] = f32x4{0.f, 0.f, 0.f, 0.f};
  }

  bf16x8 ones = bf16x8{};
  if (kgrp >= 0) {                 // B[k][0] = 1 for the bias column
#pragma unroll
    for (int j = 0; j < 8; ++j) ones[j] = (row16 == 0) ? (__bf16)1.f : (__bf16)0.f;
  }

  const long ntiles = (S / TS) * (long)B;
  for (long t = blockIdx.x; t < ntiles; t += gridDim.x) {
    const int b = (int)(t / (S / TS));
    const long s0 = (t % (S / TS)) * TS;
    for (int pp = 0; pp < NPAIR; ++pp) {
      const int p = wave + 4 * pp;
      if (p >= npairs) break;
      const int ot = p / iT, it = p - ot * iT;
      const int o = ot * 16 + row16;
      const int i = it * 16 + row16;
      const unsigned short* gr =
          (o < O) ? gz + ((long)b * O + o) * S + s0 : nullptr;
      const unsigned short* xr =
          (i < I) ? x + ((long)b * I + i) * S + s0 : nullptr;
#pragma unroll
      for (int ks = 0; ks < TS; ks += 32) {
        const long k0 = ks + kgrp * 8;
        bf16x8 a = gr ? load_frag(gr, k0) : bf16x8{};
        bf16x8 bb = xr ? load_frag(xr, k0) : bf16x8{};
        acc[pp] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bb, acc[pp], 0, 0, 0);
        if (want_bias && it == 0)
          accb[pp] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, ones, accb[pp], 0, 0, 0);
      }
    }
  }

  // writeback: D[row][col], row = (lane>>4)*4 + r, col = lane&15
#pragma unroll
  for (int pp = 0; pp < NPAIR; ++pp) {
    const int p = wave + 4 * pp;
    if (p >= npairs) continue;
    const int ot = p / iT, it = p - ot * iT;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int o = ot * 16 + kgrp * 4 + r;
      const int i = it * 16 + row16;
      if (o < O && i < I) atomicAdd(&gW[(size_t)o * I + i], acc[pp][r]);
      if (want_bias && it == 0 && o < O && row16 == 0)
        atomicAdd(&gb[o], accb[pp][r]);
    }
  }
}

int bgrid(long work) {
  long g = (work + kBlock - 1) / kBlock;
  if (g > 256L * 16) g = 256L * 16;
  if (g < 1) g = 1;
  return (int)g;
}

void check_bf16(const at::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda() && t.is_contiguous() &&
              t.scalar_type() == at::kBFloat16, name,
              " must be a contiguous CUDA bf16 tensor");
}

const unsigned short* usp(const at::Tensor& t) {
  return reinterpret_cast<const unsigned short*>(t.data_ptr());
}
unsigned short* usp_mut(at::Tensor& t) {
  return reinterpret_cast<unsigned short*>(t.data_ptr());
}

}  // namespace

std::vector<at::Tensor> bf16_channel_mix(const at::Tensor& x, const at::Tensor& W,
                                         const at::Tensor& b, bool act, bool wt,
                                         bool write_z, const at::Tensor& res) {
  check_bf16(x, "x"); check_bf16(W, "W");
  const int B = (int)x.size(0);
  const int I = (int)x.size(1);
  const long S = x.size(2);
  const int O = wt ? (int)W.size(1) : (int)W.size(0);
  TORCH_CHECK(S % kVec == 0, "bf16 channel mix: S must be a multiple of 8");
  const bool has_bias = b.defined() && b.numel() > 0;
  if (has_bias) check_bf16(b, "b");
  const bool has_res = res.defined() && res.numel() > 0;
  if (has_res) check_bf16(res, "res");

  auto y = at::empty({B, O, S}, x.options());
  auto z = write_z ? at::empty({B, O, S}, x.options()) : at::empty({0}, x.options());
  if (x.numel() == 0) return {y, z};

  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  size_t smem = sizeof(float) * ((size_t)O * I + O);
  int grid = bgrid((long)B * (S / kVec));
  const unsigned short* bp = has_bias ? usp(b) : nullptr;
  const unsigned short* rp = has_res ? usp(res) : nullptr;
  unsigned short* zp = write_z ? usp_mut(z) : nullptr;

#define BCM(KER, CAP, VEC)                                                     \
  if (act) {                                                                   \
    hipLaunchKernelGGL((KER<CAP, VEC, true>), dim3(grid), dim3(kBlock), smem,  \
                       stream, usp(x), usp(W), bp, usp_mut(y), zp, B, I, O, S, \
                       wt, has_bias, write_z, rp);                             \
  } else {                                                                     \
    hipLaunchKernelGGL((KER<CAP, VEC, false>), dim3(grid), dim3(kBlock), smem, \
                       stream, usp(x), usp(W), bp, usp_mut(y), zp, B, I, O, S, \
                       wt, has_bias, write_z, rp);                             \
  }
#define BCMP(KER, CAP, VEC, ITV, OTV)                                          \
  if (act) {                                                                   \
    hipLaunchKernelGGL((KER<CAP, VEC, true, ITV, OTV>), dim3(grid),            \
                       dim3(kBlock), smem, stream, usp(x), usp(W), bp,         \
                       usp_mut(y), zp, B, I, O, S, wt, has_bias, write_z, rp); \
  } else {                                                                     \
    hipLaunchKernelGGL((KER<CAP, VEC, false, ITV, OTV>), dim3(grid),           \
                       dim3(kBlock), smem, stream, usp(x), usp(W), bp,         \
                       usp_mut(y), zp, B, I, O, S, wt, has_bias, write_z, rp); \
  }
  // compile-time-pinned hot shapes first (block mixes, projection lift and
  // its transpose, channel lift, projection head)
  if (I == 20 && O == 20) { BCMP(bf16_channel_mix_xres_kernel, 24, 4, 20, 20) }
  else if (I == 20 && O == 128) { BCMP(bf16_channel_mix_xres_kernel, 24, 4, 20, 128) }
  else if (I == 2 && O == 20) { BCMP(bf16_channel_mix_xres_kernel, 8, 8, 2, 20) }
  else if (I == 128 && O == 20) { BCMP(bf16_channel_mix_ores_kernel, 24, 4, 128, 20) }
  else if (I == 128 && O == 1) { BCMP(bf16_channel_mix_ores_kernel, 8, 8, 128, 1) }
  else if (I <= 8) { BCM(bf16_channel_mix_xres_kernel, 8, 8) }
  else if (I <= 24) { BCM(bf16_channel_mix_xres_kernel, 24, 4) }
  else if (I <= 32) { BCM(bf16_channel_mix_xres_kernel, 32, 4) }
  else if (O <= 8) { BCMP(bf16_channel_mix_ores_kernel, 8, 8, 0, 0) }
  else if (O <= 24) { BCMP(bf16_channel_mix_ores_kernel, 24, 4, 0, 0) }
  else { TORCH_CHECK(false, "bf16 channel mix: unsupported shape I=", I, " O=", O); }
#undef BCMP
#undef BCM
  DFNO_CHECK_LAUNCH("bf16_channel_mix");
  return {y, z};
}

std::vector<at::Tensor> bf16_channel_mix_bwd_w(const at::Tensor& gz,
                                               const at::Tensor& x,
                                               bool want_bias) {
  check_bf16(gz, "gz"); check_bf16(x, "x");
  const int B = (int)x.size(0);
  const int I = (int)x.size(1);
  const long S = x.size(2);
  const int O = (int)gz.size(1);
  TORCH_CHECK(I <= 32, "bf16 grad-W: I must be <= 32");
  TORCH_CHECK(O <= 128, "bf16 grad-W: O must be <= 128");
  TORCH_CHECK(S % 128 == 0, "bf16 grad-W: S must be a multiple of 128");

  auto opts = x.options().dtype(at::kFloat);
  auto gW = at::zeros({O, I}, opts);
  auto gb = at::zeros({want_bias ? O : 0}, opts);
  if (x.numel() == 0) return {gW, gb};

  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  static const bool no_mfma = []() {
    const char* e = getenv("DFNO_BF16_GW_NO_MFMA");  // A/B knob
    return e && e[0] == '1';
  }();
  const int mfma_pairs = ((O + 15) / 16) * ((I + 15) / 16);
  if (!no_mfma && S % 256 == 0 && mfma_pairs <= 16) {
    long ntiles = (S / 256) * (long)B;
    int grid = (int)std::min(ntiles, 2048L);
    if (mfma_pairs <= 4) {
      hipLaunchKernelGGL((bf16_gw_mfma_kernel<1>), dim3(grid), dim3(kBlock), 0,
                         stream, usp(gz), usp(x), gW.data_ptr<float>(),
                         gb.numel() ? gb.data_ptr<float>() : nullptr, B, I, O,
                         S, want_bias);
    } else {
      hipLaunchKernelGGL((bf16_gw_mfma_kernel<4>), dim3(grid), dim3(kBlock), 0,
                         stream, usp(gz), usp(x), gW.data_ptr<float>(),
                         gb.numel() ? gb.data_ptr<float>() : nullptr, B, I, O,
                         S, want_bias);
    }
    DFNO_CHECK_LAUNCH("bf16_gw_mfma");
    return {gW, gb};
  }
  constexpr int LD = 132;
  const int nslab = (O + 31) / 32;
  const int o_sl = std::min(O, 32);
  size_t smem = sizeof(float) * (size_t)(32 + o_sl) * LD;
  long ntiles = (S / 128) * (long)B;
  int grid = (int)std::min(ntiles, 2048L);
  const int pairs = o_sl * I;
#define BGW(NP)                                                                \
  hipLaunchKernelGGL((bf16_gw_kernel<NP>), dim3(grid, nslab), dim3(kBlock),    \
                     smem, stream, usp(gz), usp(x), gW.data_ptr<float>(),      \
                     gb.numel() ? gb.data_ptr<float>() : nullptr, B, I, O, S,  \
                     want_bias);
  if (pairs <= 256) { BGW(1) } else if (pairs <= 512) { BGW(2) }
  else { BGW(4) }
#undef BGW
  DFNO_CHECK_LAUNCH("bf16_gw");
  return {gW, gb};
}

at::Tensor bf16_gelu_fwd(const at::Tensor& x) {
  check_bf16(x, "x");
  TORCH_CHECK(x.numel() % kVec == 0, "bf16 gelu: numel must be a multiple of 8");
  auto y = at::empty_like(x);
  if (x.numel() == 0) return y;
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  long n8 = x.numel() / kVec;
  hipLaunchKernelGGL(bf16_gelu_fwd_kernel, dim3(bgrid(n8)), dim3(kBlock), 0,
                     stream, usp(x), usp_mut(y), n8);
  DFNO_CHECK_LAUNCH("bf16_gelu");
  return y;
}

at::Tensor bf16_gelu_bwd(const at::Tensor& gy, const at::Tensor& z) {
  check_bf16(gy, "gy"); check_bf16(z, "z");
  TORCH_CHECK(gy.numel() % kVec == 0, "bf16 gelu bwd: numel % 8");
  auto gz = at::empty_like(gy);
  if (gy.numel() == 0) return gz;
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  long n8 = gy.numel() / kVec;
  hipLaunchKernelGGL(bf16_gelu_bwd_kernel, dim3(bgrid(n8)), dim3(kBlock), 0,
                     stream, usp(gy), usp(z), usp_mut(gz), n8);
  DFNO_CHECK_LAUNCH("bf16_gelu_bwd");
  return gz;
}

std::vector<at::Tensor> bf16_add_gelu(const at::Tensor& a, const at::Tensor& b) {
  check_bf16(a, "a"); check_bf16(b, "b");
  TORCH_CHECK(a.numel() == b.numel() && a.numel() % kVec == 0, "bf16 add_gelu shape");
  auto y = at::empty_like(a);
  auto z = at::empty_like(a);
  if (a.numel() == 0) return {y, z};
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  long n8 = a.numel() / kVec;
  hipLaunchKernelGGL(bf16_add_gelu_kernel, dim3(bgrid(n8)), dim3(kBlock), 0,
                     stream, usp(a), usp(b), usp_mut(y), usp_mut(z), n8);
  DFNO_CHECK_LAUNCH("bf16_add_gelu");
  return {y, z};
}
