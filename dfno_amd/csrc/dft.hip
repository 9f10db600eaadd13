// gfx950 fused truncated-spectrum DFT kernels (see ops/fft.py for the
// semantics and the roofline argument).
//
// For FNO mode counts (m ~ 8..32 kept of N <= 64) a truncated naive DFT is
// N*m complex MACs per line — FLOP-comparable to a full FFT — and lets one
// kernel fuse transform + truncation/zero-padding + 1/n scale while reading
// the tensor once along its NATIVE strides (no transposes, no hipFFT
// staging copies, no separate cat/zeros passes, no autograd mirror copies).
//
// Geometry: tensor viewed as [outer, L, inner] around the transform dim.
//  * inner > 1 (the spatial dims): thread <-> (outer, inner) line; loads of
//    x[o, j, :] are lane-consecutive in `inner` -> fully coalesced.
//  * inner == 1 (the trailing rfft/irfft time dim): each thread streams its
//    own contiguous line; a wave's combined footprint is a contiguous span,
//    so L1/L2 serve the per-thread scalar loads after the first touch, and
//    per-line outputs are >= 64B contiguous runs.
//
// Twiddles: one table of w^r = exp(-2*pi*i*r/N), r in [0, N), built in LDS
// at block start (sign applied on read); (j*k) mod N tracked incrementally
// per kept mode in registers (unrolled + predicated, no scratch).

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <torch/extension.h>
#include <mutex>
#include <unordered_map>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>

#include "kernels.h"

namespace {

constexpr int kBlock = 256;
constexpr int kMaxN = 64;    // cap of the fully-tuned N<=64 paths
constexpr int kMaxNBig = 256;  // cap of the generalized radix-8xNB / LB paths

template <typename T>
__device__ __forceinline__ void sincos_t(T a, T* s, T* c);
template <>
__device__ __forceinline__ void sincos_t<float>(float a, float* s, float* c) {
  sincosf(a, s, c);
}
template <>
__device__ __forceinline__ void sincos_t<double>(double a, double* s, double* c) {
  *s = sin(a); *c = cos(a);
}

// Per-mode twiddle recurrence: each kept mode ki keeps its current factor
// w^{j k(ki)} in two registers and multiplies by a per-mode step each j —
// pure independent fma chains that pipeline fully (an LDS table costs a
// dependent ds_read per MAC and stalls ~10x).  Drift over N <= 64 steps is
// O(N eps), far below the fp32/fp64 tolerances used here.

template <typename T>
__device__ __forceinline__ void cmul_acc(T& cr, T& ci, T sr, T si) {
  T nr = cr * sr - ci * si;
  T ni = cr * si + ci * sr;
  cr = nr; ci = ni;
}

// bf16 <-> f32 helpers for the bf16-IO transform variants (the bf16 model's
// real-side activations enter/leave the spectral path in bf16 storage;
// arithmetic stays fp32 — see csrc/bf16.hip for the rationale)
__device__ __forceinline__ float dft_b2f(unsigned short h) {
  return __uint_as_float(((unsigned int)h) << 16);
}

__device__ __forceinline__ unsigned int dft_f2b2(float lo, float hi) {
  __hip_bfloat162 h2 = __float22bfloat162_rn(float2{lo, hi});
  return *reinterpret_cast<unsigned int*>(&h2);
}

// kept-mode k value for index ki
__device__ __forceinline__ int kept_k(int ki, int m_lo, int N, int m_hi) {
  return (ki < m_lo) ? ki : (N - m_hi + (ki - m_lo));
}

// ---------------------------------------------------------------------------
// C2C analysis: out[o, ki, i] = scale * sum_j in[o, j, i] * w^{-j k(ki)}
// ---------------------------------------------------------------------------

template <typename T, int MCAP>
__global__ __launch_bounds__(kBlock) void dft_c2c_analysis_kernel(
    const T* __restrict__ in, T* __restrict__ out,
    long outer, int N, long inner, int m_lo, int m_hi, T scale) {
  const int m = m_lo + m_hi;
  T str[MCAP], sti[MCAP];
#pragma unroll
  for (int ki = 0; ki < MCAP; ++ki) {
    if (ki < m) {
      int k = kept_k(ki, m_lo, N, m_hi);
      sincos_t<T>(T(-2.0) * T(M_PI) * T(k) / T(N), &sti[ki], &str[ki]);
    }
  }

  long total = outer * inner;
  long t0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long t = t0; t < total; t += stride) {
    long o = t / inner;
    long i = t % inner;
    const T* src = in + 2 * (o * N * inner + i);

    T ar[MCAP], ai[MCAP], cr[MCAP], ci[MCAP];
#pragma unroll
    for (int ki = 0; ki < MCAP; ++ki) {
      if (ki < m) { ar[ki] = T(0); ai[ki] = T(0); cr[ki] = T(1); ci[ki] = T(0); }
    }
    for (int j = 0; j < N; ++j) {
      const T xr = src[2 * j * inner];
      const T xi = src[2 * j * inner + 1];
#pragma unroll
      for (int ki = 0; ki < MCAP; ++ki) {
        if (ki < m) {
          ar[ki] += xr * cr[ki] - xi * ci[ki];
          ai[ki] += xr * ci[ki] + xi * cr[ki];
          cmul_acc(cr[ki], ci[ki], str[ki], sti[ki]);
        }
      }
    }
    T* dst = out + 2 * (o * m * inner + i);
#pragma unroll
    for (int ki = 0; ki < MCAP; ++ki) {
      if (ki < m) {
        dst[2 * ki * inner] = scale * ar[ki];
        dst[2 * ki * inner + 1] = scale * ai[ki];
      }
    }
  }
}

// ---------------------------------------------------------------------------
// C2C synthesis: out[o, j, i] = scale * sum_ki in[o, ki, i] * w^{+j k(ki)}
// ---------------------------------------------------------------------------

template <typename T, int MCAP>
__global__ __launch_bounds__(kBlock) void dft_c2c_synthesis_kernel(
    const T* __restrict__ in, T* __restrict__ out,
    long outer, int N, long inner, int m_lo, int m_hi, T scale) {
  const int m = m_lo + m_hi;
  T str[MCAP], sti[MCAP];
#pragma unroll
  for (int ki = 0; ki < MCAP; ++ki) {
    if (ki < m) {
      int k = kept_k(ki, m_lo, N, m_hi);
      sincos_t<T>(T(2.0) * T(M_PI) * T(k) / T(N), &sti[ki], &str[ki]);
    }
  }

  long total = outer * inner;
  long t0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long t = t0; t < total; t += stride) {
    long o = t / inner;
    long i = t % inner;
    const T* src = in + 2 * (o * m * inner + i);

    T yr[MCAP], yi[MCAP], cr[MCAP], ci[MCAP];
#pragma unroll
    for (int ki = 0; ki < MCAP; ++ki) {
      if (ki < m) {
        yr[ki] = scale * src[2 * ki * inner];
        yi[ki] = scale * src[2 * ki * inner + 1];
        cr[ki] = T(1); ci[ki] = T(0);
      }
    }
    T* dst = out + 2 * (o * N * inner + i);
    for (int j = 0; j < N; ++j) {
      T sr = T(0), si = T(0);
#pragma unroll
      for (int ki = 0; ki < MCAP; ++ki) {
        if (ki < m) {
          sr += yr[ki] * cr[ki] - yi[ki] * ci[ki];
          si += yr[ki] * ci[ki] + yi[ki] * cr[ki];
          cmul_acc(cr[ki], ci[ki], str[ki], sti[ki]);
        }
      }
      dst[2 * j * inner] = sr;
      dst[2 * j * inner + 1] = si;
    }
  }
}

// ---------------------------------------------------------------------------
// Paired/vectorized C2C variants: when m_hi == m_lo (the FNO case), the
// suffix mode N-k has twiddle conj(w^{-jk}), so both directions share one
// recurrence chain per prefix k (plus one for the unpaired k = m_lo); and
// two adjacent `inner` elements are processed per thread (float4 loads,
// twiddle work amortized): ~1.6x fewer VALU ops per element.
// ---------------------------------------------------------------------------

// G-way j-split: G threads (same wave) share one pair's j-loop, each
// walking j = g, g+G, ... with chain step w^{-kG} and start w^{-kg}; the
// partial sums reduce with log2(G) shfl_xor steps.  Needs G | N so the
// chains wrap to their start after each line (w^{-kN} = 1).  Raises wave
// occupancy G-fold for the middle-dim transforms whose outer*pairs alone
// underfills 1024 SIMDs (the x/y-dim c2c calls of the flagship run at 0.2-
// 0.5 waves/SIMD otherwise).
template <typename T, int LCAP, int G = 1>
__global__ __launch_bounds__(kBlock) void dft_c2c_analysis2_kernel(
    const T* __restrict__ in, T* __restrict__ out, const T* __restrict__ tw,
    long outer, int N, long inner, int m_lo, T scale) {
  // m_hi == m_lo; tw is the [N, m_lo+1, 2] table of w^{-jk} (see r2c note:
  // register twiddle chains were 4 of every ~5 non-MAC VALU ops and cost 4
  // LCAP register arrays; (j,k) are lane-uniform so table reads scalarize)
  const int nch = m_lo + 1;
  constexpr int PL = 64 / G;           // pairs handled per wave per g-group
  const int lane = (int)(threadIdx.x % 64);
  const int g = lane / PL;

  long pairs = inner / 2;
  long total = outer * pairs;
  long p0 = (((long)blockIdx.x * blockDim.x + threadIdx.x) / 64) * PL +
            (lane % PL);
  long pstride = ((long)gridDim.x * blockDim.x / 64) * PL;
  const int iters = N / G;             // host guarantees G | N
  for (long t = p0; t < total; t += pstride) {
    long o = t / pairs;
    long i = (t % pairs) * 2;
    const T* src = in + 2 * (o * N * inner + i);

    // acc[ki][elem]{r,i}; ki < 2*m_lo
    T a0r[LCAP], a0i[LCAP], a1r[LCAP], a1i[LCAP];   // prefix accs (elem0/1)
    T b0r[LCAP], b0i[LCAP], b1r[LCAP], b1i[LCAP];   // suffix accs
#pragma unroll
    for (int k = 0; k < LCAP; ++k) {
      if (k < m_lo) {
        a0r[k] = a0i[k] = a1r[k] = a1i[k] = T(0);
        b0r[k] = b0i[k] = b1r[k] = b1i[k] = T(0);
      }
    }
    // software pipeline: the next j's load issues BEFORE this j's ~200-fma
    // block (the scheduler will not hoist it past that body on its own;
    // measured 6.3 TB/s for a bare read loop vs 1.9 here without this)
    float4 v;
    if constexpr (std::is_same<T, float>::value)
      v = *reinterpret_cast<const float4*>(src + 2 * g * inner);
    for (int sct = 0; sct < iters; ++sct) {
      const int j = g + sct * G;
      T x0r, x0i, x1r, x1i;
      if constexpr (std::is_same<T, float>::value) {
        x0r = v.x; x0i = v.y; x1r = v.z; x1i = v.w;
        if (sct + 1 < iters)
          v = *reinterpret_cast<const float4*>(src + 2 * (j + G) * inner);
      } else {
        x0r = src[2 * j * inner];
        x0i = src[2 * j * inner + 1];
        x1r = src[2 * j * inner + 2];
        x1i = src[2 * j * inner + 3];
      }
      auto twj = (const __attribute__((address_space(4))) T*)
          (tw + (long)j * 2 * nch);
#pragma unroll
      for (int k = 0; k < LCAP; ++k) {
        if (k < m_lo) {
          // prefix mode k: w^{-jk} = (cr, ci)
          const T cr = twj[2 * k], ci = twj[2 * k + 1];
          a0r[k] += x0r * cr - x0i * ci;
          a0i[k] += x0r * ci + x0i * cr;
          a1r[k] += x1r * cr - x1i * ci;
          a1i[k] += x1r * ci + x1i * cr;
          // suffix mode N - k' where k' = m_lo - k: uses conj(w^{-jk'})
          const int kp = m_lo - k;
          const T dr = twj[2 * kp], di = -twj[2 * kp + 1];
          b0r[k] += x0r * dr - x0i * di;
          b0i[k] += x0r * di + x0i * dr;
          b1r[k] += x1r * dr - x1i * di;
          b1i[k] += x1r * di + x1i * dr;
        }
      }
    }
    if constexpr (G > 1) {
      // combine the G partial sums (partners share (o, pair), differ in g)
#pragma unroll
      for (int k = 0; k < LCAP; ++k) {
        if (k < m_lo) {
          for (int off = PL; off < 64; off <<= 1) {
            a0r[k] += __shfl_xor(a0r[k], off, 64);
            a0i[k] += __shfl_xor(a0i[k], off, 64);
            a1r[k] += __shfl_xor(a1r[k], off, 64);
            a1i[k] += __shfl_xor(a1i[k], off, 64);
            b0r[k] += __shfl_xor(b0r[k], off, 64);
            b0i[k] += __shfl_xor(b0i[k], off, 64);
            b1r[k] += __shfl_xor(b1r[k], off, 64);
            b1i[k] += __shfl_xor(b1i[k], off, 64);
          }
        }
      }
      if (g != 0) continue;            // one g-group writes
    }
    const int m = 2 * m_lo;
    T* dst = out + 2 * (o * m * inner + i);
#pragma unroll
    for (int k = 0; k < LCAP; ++k) {
      if (k < m_lo) {
        // prefix output ki = k
        dst[2 * k * inner] = scale * a0r[k];
        dst[2 * k * inner + 1] = scale * a0i[k];
        dst[2 * k * inner + 2] = scale * a1r[k];
        dst[2 * k * inner + 3] = scale * a1i[k];
        // suffix output ki = m_lo + k corresponds to kglobal = N - (m_lo - k)
        int ks = m_lo + k;
        dst[2 * ks * inner] = scale * b0r[k];
        dst[2 * ks * inner + 1] = scale * b0i[k];
        dst[2 * ks * inner + 2] = scale * b1r[k];
        dst[2 * ks * inner + 3] = scale * b1i[k];
      }
    }
  }
}

// G-way j-split synthesis: outputs j = g, g+G, ... are independent, so no
// reduction is needed; each g-group loads the m kept modes (same lines, L1
// broadcast) and writes its own j's.  Needs G | N for the chain wrap.
template <typename T, int LCAP, int G = 1>
__global__ __launch_bounds__(kBlock) void dft_c2c_synthesis2_kernel(
    const T* __restrict__ in, T* __restrict__ out, const T* __restrict__ tw,
    long outer, int N, long inner, int m_lo, T scale) {
  // tw is the [N, m_lo+1, 2] table of w^{+jk}; see analysis2 note
  const int nch = m_lo + 1;
  constexpr int PL = 64 / G;
  const int lane = (int)(threadIdx.x % 64);
  const int g = lane / PL;

  const int m = 2 * m_lo;
  long pairs = inner / 2;
  long total = outer * pairs;
  long p0 = (((long)blockIdx.x * blockDim.x + threadIdx.x) / 64) * PL +
            (lane % PL);
  long pstride = ((long)gridDim.x * blockDim.x / 64) * PL;
  const int iters = N / G;             // host guarantees G | N
  for (long t = p0; t < total; t += pstride) {
    long o = t / pairs;
    long i = (t % pairs) * 2;
    const T* src = in + 2 * (o * m * inner + i);

    T p0r[LCAP], p0i[LCAP], p1r[LCAP], p1i[LCAP];   // prefix inputs
    T q0r[LCAP], q0i[LCAP], q1r[LCAP], q1i[LCAP];   // suffix inputs
#pragma unroll
    for (int k = 0; k < LCAP; ++k) {
      if (k < m_lo) {
        p0r[k] = scale * src[2 * k * inner];
        p0i[k] = scale * src[2 * k * inner + 1];
        p1r[k] = scale * src[2 * k * inner + 2];
        p1i[k] = scale * src[2 * k * inner + 3];
        int ks = m_lo + k;
        q0r[k] = scale * src[2 * ks * inner];
        q0i[k] = scale * src[2 * ks * inner + 1];
        q1r[k] = scale * src[2 * ks * inner + 2];
        q1i[k] = scale * src[2 * ks * inner + 3];
      }
    }
    T* dst = out + 2 * (o * N * inner + i);
    for (int sct = 0; sct < iters; ++sct) {
      const int j = g + sct * G;
      T s0r = T(0), s0i = T(0), s1r = T(0), s1i = T(0);
      auto twj = (const __attribute__((address_space(4))) T*)
          (tw + (long)j * 2 * nch);
#pragma unroll
      for (int k = 0; k < LCAP; ++k) {
        if (k < m_lo) {
          // prefix mode k with w^{+jk} = (cr, ci)
          const T cr = twj[2 * k], ci = twj[2 * k + 1];
          s0r += p0r[k] * cr - p0i[k] * ci;
          s0i += p0r[k] * ci + p0i[k] * cr;
          s1r += p1r[k] * cr - p1i[k] * ci;
          s1i += p1r[k] * ci + p1i[k] * cr;
          // suffix kglobal = N - (m_lo - k): w^{+j(N-kp)} = conj(w^{+jkp})
          const int kp = m_lo - k;
          const T dr = twj[2 * kp], di = -twj[2 * kp + 1];
          s0r += q0r[k] * dr - q0i[k] * di;
          s0i += q0r[k] * di + q0i[k] * dr;
          s1r += q1r[k] * dr - q1i[k] * di;
          s1i += q1r[k] * di + q1i[k] * dr;
        }
      }
      if constexpr (std::is_same<T, float>::value) {
        *reinterpret_cast<float4*>(dst + 2 * j * inner) =
            make_float4(s0r, s0i, s1r, s1i);
      } else {
        dst[2 * j * inner] = s0r;
        dst[2 * j * inner + 1] = s0i;
        dst[2 * j * inner + 2] = s1r;
        dst[2 * j * inner + 3] = s1i;
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Radix-8x8 C2C for N == 64 (every paired middle-dim transform of the
// flagship): j = 8a + b.  Stage 1 is a constant-twiddle 8-point DFT over a
// for each b (radix-2 butterflies, ~60 real ops); stage 2 accumulates the
// kept modes with w64^{bk} = tw[b][k] (the first 8 rows of the existing
// twiddle table).  Cuts VALU work ~4.7x vs the naive m*N loop, which PMC
// showed is the dominant cost (these kernels run at 25-50% VALU issue).
// ---------------------------------------------------------------------------

// 8-point DFT of (xr,xi)[8]: A[r] = sum_a x[a] e^{SGN 2pi i a r / 8}
template <typename T, int SGN>
__device__ __forceinline__ void dft8(const T* xr, const T* xi, T* Ar, T* Ai) {
  const T C = T(0.7071067811865476);
  // 4-point on evens (0,2,4,6) and odds (1,3,5,7)
  T E0r, E0i, E1r, E1i, E2r, E2i, E3r, E3i;
  T O0r, O0i, O1r, O1i, O2r, O2i, O3r, O3i;
  {
    T pr = xr[0] + xr[4], pi = xi[0] + xi[4];
    T qr = xr[0] - xr[4], qi = xi[0] - xi[4];
    T tr = xr[2] + xr[6], ti = xi[2] + xi[6];
    T dr = xr[2] - xr[6], di = xi[2] - xi[6];
    T sr = (SGN < 0) ? di : -di;          // d * (SGN i)^... = -+i d
    T si = (SGN < 0) ? -dr : dr;
    E0r = pr + tr; E0i = pi + ti;
    E1r = qr + sr; E1i = qi + si;
    E2r = pr - tr; E2i = pi - ti;
    E3r = qr - sr; E3i = qi - si;
  }
  {
    T pr = xr[1] + xr[5], pi = xi[1] + xi[5];
    T qr = xr[1] - xr[5], qi = xi[1] - xi[5];
    T tr = xr[3] + xr[7], ti = xi[3] + xi[7];
    T dr = xr[3] - xr[7], di = xi[3] - xi[7];
    T sr = (SGN < 0) ? di : -di;
    T si = (SGN < 0) ? -dr : dr;
    O0r = pr + tr; O0i = pi + ti;
    O1r = qr + sr; O1i = qi + si;
    O2r = pr - tr; O2i = pi - ti;
    O3r = qr - sr; O3i = qi - si;
  }
  // A[r] = E[r] + w8^{SGN r} O[r]; A[r+4] = E[r] - w8^{SGN r} O[r]
  Ar[0] = E0r + O0r; Ai[0] = E0i + O0i;
  Ar[4] = E0r - O0r; Ai[4] = E0i - O0i;
  {
    // w8^{SGN 1} = C (1 + SGN i)
    T wr = C * (O1r - T(SGN) * O1i);
    T wi = C * (O1i + T(SGN) * O1r);
    Ar[1] = E1r + wr; Ai[1] = E1i + wi;
    Ar[5] = E1r - wr; Ai[5] = E1i - wi;
  }
  {
    // w8^{SGN 2} = SGN i
    T wr = -T(SGN) * O2i;
    T wi = T(SGN) * O2r;
    Ar[2] = E2r + wr; Ai[2] = E2i + wi;
    Ar[6] = E2r - wr; Ai[6] = E2i - wi;
  }
  {
    // w8^{SGN 3} = C (-1 + SGN i)
    T wr = C * (-O3r - T(SGN) * O3i);
    T wi = C * (-O3i + T(SGN) * O3r);
    Ar[3] = E3r + wr; Ai[3] = E3i + wi;
    Ar[7] = E3r - wr; Ai[7] = E3i - wi;
  }
}

// Generalized to N = 8 * NBT via decimation j = NBT*a + b (a in [0,8),
// b in [0,NBT)): w_N^{(NBT a + b)k} = w_8^{a (k mod 8)} * w_N^{bk}, so
// stage 1 is an 8-point DFT over the stride-NBT comb of each b and stage 2
// accumulates kept modes with the first NBT rows of the [N, m_lo+1] table.
// NBT=8 is the round-1 N=64 kernel; 16/32 cover the 128/256 grids of the
// weak-scaling configs (reference gen_scripts.py:140-153) natively.
template <typename T, int LCAP, int MLT = 0, int NBT = 8>
__global__ __launch_bounds__(kBlock) void dft_c2c_radix8_ana_kernel(
    const T* __restrict__ in, T* __restrict__ out, const T* __restrict__ tw,
    long outer, long inner, int m_lo_, T scale) {
  // MLT > 0 pins m_lo at compile time (folded table strides / no
  // predicates; see the r2c fast-path note)
  const int m_lo = MLT > 0 ? MLT : m_lo_;
  constexpr int N = 8 * NBT;
  const int nch = m_lo + 1;
  long pairs = inner / 2;
  long total = outer * pairs;
  long t0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long t = t0; t < total; t += stride) {
    long o = t / pairs;
    long i = (t % pairs) * 2;
    const T* src = in + 2 * (o * N * inner + i);

    T a0r[LCAP], a0i[LCAP], a1r[LCAP], a1i[LCAP];
    T b0r[LCAP], b0i[LCAP], b1r[LCAP], b1i[LCAP];
#pragma unroll
    for (int k = 0; k < LCAP; ++k) {
      if (k < m_lo) {
        a0r[k] = a0i[k] = a1r[k] = a1i[k] = T(0);
        b0r[k] = b0i[k] = b1r[k] = b1i[k] = T(0);
      }
    }
    for (int b = 0; b < NBT; ++b) {
      T x0r[8], x0i[8], x1r[8], x1i[8];
#pragma unroll
      for (int a = 0; a < 8; ++a) {
        const float4 v = *reinterpret_cast<const float4*>(
            src + 2 * (NBT * a + b) * inner);
        x0r[a] = v.x; x0i[a] = v.y; x1r[a] = v.z; x1i[a] = v.w;
      }
      T A0r[8], A0i[8], A1r[8], A1i[8];
      dft8<T, -1>(x0r, x0i, A0r, A0i);
      dft8<T, -1>(x1r, x1i, A1r, A1i);
      auto twb = (const __attribute__((address_space(4))) T*)
          (tw + (long)b * 2 * nch);
#pragma unroll
      for (int k = 0; k < LCAP; ++k) {
        if (k < m_lo) {
          // prefix mode k: A[k mod 8] * w64^{bk}
          const int r = k & 7;
          const T cr = twb[2 * k], ci = twb[2 * k + 1];
          a0r[k] += A0r[r] * cr - A0i[r] * ci;
          a0i[k] += A0r[r] * ci + A0i[r] * cr;
          a1r[k] += A1r[r] * cr - A1i[r] * ci;
          a1i[k] += A1r[r] * ci + A1i[r] * cr;
          // suffix mode N - kp (kp = m_lo - k): r = (-kp) mod 8,
          // w64^{b(N-kp)} = conj(w64^{b kp})
          const int kp = m_lo - k;
          const int rs = (8 - (kp & 7)) & 7;
          const T dr = twb[2 * kp], di = -twb[2 * kp + 1];
          b0r[k] += A0r[rs] * dr - A0i[rs] * di;
          b0i[k] += A0r[rs] * di + A0i[rs] * dr;
          b1r[k] += A1r[rs] * dr - A1i[rs] * di;
          b1i[k] += A1r[rs] * di + A1i[rs] * dr;
        }
      }
    }
    const int m = 2 * m_lo;
    T* dst = out + 2 * (o * m * inner + i);
#pragma unroll
    for (int k = 0; k < LCAP; ++k) {
      if (k < m_lo) {
        dst[2 * k * inner] = scale * a0r[k];
        dst[2 * k * inner + 1] = scale * a0i[k];
        dst[2 * k * inner + 2] = scale * a1r[k];
        dst[2 * k * inner + 3] = scale * a1i[k];
        int ks = m_lo + k;
        dst[2 * ks * inner] = scale * b0r[k];
        dst[2 * ks * inner + 1] = scale * b0i[k];
        dst[2 * ks * inner + 2] = scale * b1r[k];
        dst[2 * ks * inner + 3] = scale * b1i[k];
      }
    }
  }
}

// Radix-8xNB synthesis: X[NBT a + b] = sum_r w8^{+ar} C_b[r] with C_b[r] the
// per-residue accumulation of the kept modes times w_N^{+bk} (prefix) /
// conj(w_N^{+b kp}) (suffix).  tw here is the synthesis-signed table.
template <typename T, int LCAP, int MLT = 0, int NBT = 8>
__global__ __launch_bounds__(kBlock) void dft_c2c_radix8_syn_kernel(
    const T* __restrict__ in, T* __restrict__ out, const T* __restrict__ tw,
    long outer, long inner, int m_lo_, T scale) {
  // MLT > 0 pins m_lo at compile time (folded table strides / no
  // predicates; see the r2c fast-path note)
  const int m_lo = MLT > 0 ? MLT : m_lo_;
  constexpr int N = 8 * NBT;
  const int nch = m_lo + 1;
  const int m = 2 * m_lo;
  long pairs = inner / 2;
  long total = outer * pairs;
  long t0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long t = t0; t < total; t += stride) {
    long o = t / pairs;
    long i = (t % pairs) * 2;
    const T* src = in + 2 * (o * m * inner + i);

    T p0r[LCAP], p0i[LCAP], p1r[LCAP], p1i[LCAP];
    T q0r[LCAP], q0i[LCAP], q1r[LCAP], q1i[LCAP];
#pragma unroll
    for (int k = 0; k < LCAP; ++k) {
      if (k < m_lo) {
        const float4 v = *reinterpret_cast<const float4*>(src + 2 * k * inner);
        p0r[k] = scale * v.x; p0i[k] = scale * v.y;
        p1r[k] = scale * v.z; p1i[k] = scale * v.w;
        int ks = m_lo + k;
        const float4 w = *reinterpret_cast<const float4*>(src + 2 * ks * inner);
        q0r[k] = scale * w.x; q0i[k] = scale * w.y;
        q1r[k] = scale * w.z; q1i[k] = scale * w.w;
      }
    }
    T* dst = out + 2 * (o * N * inner + i);
    auto syn_b = [&](int b) {
      T C0r[8], C0i[8], C1r[8], C1i[8];
#pragma unroll
      for (int r = 0; r < 8; ++r) { C0r[r] = C0i[r] = C1r[r] = C1i[r] = T(0); }
      auto twb = (const __attribute__((address_space(4))) T*)
          (tw + (long)b * 2 * nch);
#pragma unroll
      for (int k = 0; k < LCAP; ++k) {
        if (k < m_lo) {
          const int r = k & 7;
          const T cr = twb[2 * k], ci = twb[2 * k + 1];
          C0r[r] += p0r[k] * cr - p0i[k] * ci;
          C0i[r] += p0r[k] * ci + p0i[k] * cr;
          C1r[r] += p1r[k] * cr - p1i[k] * ci;
          C1i[r] += p1r[k] * ci + p1i[k] * cr;
          const int kp = m_lo - k;
          const int rs = (8 - (kp & 7)) & 7;
          const T dr = twb[2 * kp], di = -twb[2 * kp + 1];
          C0r[rs] += q0r[k] * dr - q0i[k] * di;
          C0i[rs] += q0r[k] * di + q0i[k] * dr;
          C1r[rs] += q1r[k] * dr - q1i[k] * di;
          C1i[rs] += q1r[k] * di + q1i[k] * dr;
        }
      }
      T X0r[8], X0i[8], X1r[8], X1i[8];
      dft8<T, 1>(C0r, C0i, X0r, X0i);
      dft8<T, 1>(C1r, C1i, X1r, X1i);
#pragma unroll
      for (int a = 0; a < 8; ++a) {
        *reinterpret_cast<float4*>(dst + 2 * (NBT * a + b) * inner) =
            make_float4(X0r[a], X0i[a], X1r[a], X1i[a]);
      }
    };
    if constexpr (NBT == 8) {
      for (int b = 0; b < 8; ++b) syn_b(b);
    } else {
      // cap the b-unroll: full unroll at NBT=16/32 runs the register file
      // to 256 VGPR (1 wave/SIMD) with no twiddle-fold benefit
#pragma clang loop unroll_count(2)
      for (int b = 0; b < NBT; ++b) syn_b(b);
    }
  }
}

// ---------------------------------------------------------------------------
// R2C (last dim): out[l, k] = fac_k * scale * sum_j in[l, j] * w^{-jk}
// ---------------------------------------------------------------------------

// glds double-buffered variant (fp32): the cooperative load->ds_write stage
// serializes against the compute phase (probe: full 0.40 ms vs 0.19 ms for
// stage+write alone at the flagship shape; phase-skewing co-resident blocks
// changes nothing).  Staging tile t+1 by global_load_lds DMA while t is
// being consumed, with a counted s_waitcnt vmcnt(NG) + raw barriers, runs
// the same shape at 5.4 TB/s -- 2.3x the staged version (r2cprobe.hip).
// TI = unsigned short DMAs the raw bf16 tile (half the LDS and HBM bytes)
// and converts at the compute read; T stays the fp32 math type.
template <typename T, int MCAP, int NG, int NT = 0, typename TI = T>
__global__ __launch_bounds__(kBlock) void dft_r2c_glds_kernel(
    const TI* __restrict__ in, T* __restrict__ out, const T* __restrict__ tw,
    long lines, int N_, int m, T scale, bool factors) {
  // NT > 0 pins the transform length AND the mode count (m == MCAP, host-
  // checked) at compile time: the j-loop fully unrolls with all twiddle
  // offsets folded to immediates, batching the s_loads and exposing the fma
  // ILP.  With runtime N/m the loop serializes on per-iteration scalar-load
  // waits (0.42 ms) or, unrolled with runtime-m addressing, spills SGPRs;
  // the folded version runs 0.18 ms at the flagship shape (r2cprobe.hip).
  const int N = NT > 0 ? NT : N_;
  const int mm = NT > 0 ? MCAP : m;
  if constexpr (!std::is_same<T, float>::value) return;  // float-math path
  constexpr int EPV = 16 / (int)sizeof(TI);          // elements per 16B DMA
  extern __shared__ __align__(16) char smem_raw[];
  TI* ring = reinterpret_cast<TI*>(smem_raw);        // [2][kBlock * N]
  const int tilef = kBlock * N;                      // elements per tile
  // the counted s_waitcnt needs a compile-time-exact per-wave instruction
  // count: host picks NG = ceil(N/4) rounded up to a supported value and
  // issues are padded to exactly NG glds (pad slots re-stage the tile's
  // last 16 bytes; a whole-wave all-lanes-same-address pad instruction is
  // NOT free, so NG tracks N instead of a single worst case)
  const int wave = (int)(threadIdx.x / 64), lane = (int)(threadIdx.x % 64);
  const long nfl = lines * N;
  long ntiles = (lines + kBlock - 1) / kBlock;

  auto issue = [&](int buf, long tb) {
    const TI* src = in;
    const long base = tb * (long)tilef;
    TI* dst = ring + (long)buf * tilef;
#pragma unroll
    for (int k = 0; k < NG; ++k) {
      long fo = (long)(wave * 64 + k * kBlock + lane) * EPV;
      if (fo + EPV > tilef) fo = tilef - EPV;        // pad: clamp in-tile
      long gfo = base + fo;
      if (gfo + EPV > nfl) gfo = nfl - EPV;          // clamp last tile
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)(src + gfo),
          (__attribute__((address_space(3))) void*)&dst[fo], 16, 0, 0);
    }
  };

  // blocks with no tile must not issue: an outstanding LDS-DMA at kernel
  // exit (nothing ever waits on it) is undefined once the LDS is reassigned
  if (blockIdx.x < ntiles) issue(0, blockIdx.x);
  long c = 0;
  for (long tb = blockIdx.x; tb < ntiles; tb += gridDim.x, ++c) {
    long nxt = tb + gridDim.x;
    if (nxt < ntiles) {
      issue((int)((c + 1) & 1), nxt);
      asm volatile("s_waitcnt vmcnt(%0)" ::"n"(NG) : "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();
    asm volatile("" ::: "memory");

    const TI* tile = ring + (long)(c & 1) * tilef;
    const TI* src = tile + threadIdx.x * N;
    T ar[MCAP], ai[MCAP];
#pragma unroll
    for (int k = 0; k < MCAP; ++k) {
      if (k < mm) { ar[k] = T(0); ai[k] = T(0); }
    }
#pragma unroll
    for (int j = 0; j < (NT > 0 ? NT : 64); ++j) {
      if (NT == 0 && j >= N) break;
      T x;
      if constexpr (std::is_same<TI, unsigned short>::value)
        x = dft_b2f(src[j]);
      else
        x = (T)src[j];
      auto twj = (const __attribute__((address_space(4))) T*)
          (tw + (long)(j * 2) * mm);
#pragma unroll
      for (int k = 0; k < MCAP; ++k) {
        if (k < mm) {
          ar[k] += x * twj[2 * k];
          ai[k] += x * twj[2 * k + 1];
        }
      }
    }
    long l0 = tb * kBlock;
    if (l0 + (long)threadIdx.x < lines) {
      T* dst = out + 2 * (l0 + threadIdx.x) * mm;
#pragma unroll
      for (int k = 0; k < MCAP; ++k) {
        if (k < mm) {
          T f = T(1);
          bool edge = (k == 0) || (N % 2 == 0 && 2 * k == N);
          if (factors && !edge) f = T(2);
          dst[2 * k] = f * scale * ar[k];
          dst[2 * k + 1] = (factors && edge) ? T(0) : f * scale * ai[k];
        }
      }
    }
    // all waves out of ring[c&1] before its re-issue next iteration
    __builtin_amdgcn_s_barrier();
    asm volatile("" ::: "memory");
  }
}

// One 256-line tile is cooperatively staged into LDS with coalesced loads
// (per-thread direct line reads thrash L1: 16 waves x line-span > 32 KiB,
// measured SQ_WAIT_ANY = 72%); threads then stream their line from LDS.
// Twiddles come from a host-precomputed [N, m, 2] table instead of per-mode
// register recurrences: the chain update was 4 of every 6 VALU ops (PMC:
// VALU-bound at ~85% issue), and (j, k) are lane-uniform so the table reads
// compile to scalar loads that the k-unrolled fma stream hides entirely.
// TI != T stages a bf16 input (convert at the LDS write; compute stays T)
template <typename T, int MCAP, int LB = kBlock, typename TI = T>
__global__ __launch_bounds__(LB) void dft_r2c_last_kernel(
    const TI* __restrict__ in, T* __restrict__ out, const T* __restrict__ tw,
    long lines, int N, int m, T scale, bool factors) {
  // LB = block size = lines per tile; 128/64 for N = 128/256 keep the
  // [LB * N] LDS tile within the 160 KiB budget (2 blocks/CU at 64 KiB).
  constexpr bool kBf16In = std::is_same<TI, unsigned short>::value;
  extern __shared__ __align__(16) char smem_raw[];
  T* tile = reinterpret_cast<T*>(smem_raw);   // [LB * N]

  long ntiles = (lines + LB - 1) / LB;
  for (long tb = blockIdx.x; tb < ntiles; tb += gridDim.x) {
    long l0 = tb * LB;
    int nl = (int)min((long)LB, lines - l0);
    __syncthreads();
    if constexpr (kBf16In) {
      const long base = l0 * N;
      if ((nl * N) % 8 == 0 && (base % 8 == 0) &&
          ((reinterpret_cast<uintptr_t>(in) & 15) == 0)) {
        for (int idx = threadIdx.x * 8; idx < nl * N; idx += LB * 8) {
          const uint4 raw = *reinterpret_cast<const uint4*>(in + base + idx);
          const unsigned int w[4] = {raw.x, raw.y, raw.z, raw.w};
#pragma unroll
          for (int k = 0; k < 4; ++k) {
            tile[idx + 2 * k] = (T)dft_b2f((unsigned short)(w[k] & 0xffffu));
            tile[idx + 2 * k + 1] = (T)dft_b2f((unsigned short)(w[k] >> 16));
          }
        }
      } else {
        for (int idx = threadIdx.x; idx < nl * N; idx += LB)
          tile[idx] = (T)dft_b2f(in[base + idx]);
      }
    } else if constexpr (std::is_same<T, float>::value) {
      const long base = l0 * N;
      if ((nl * N) % 4 == 0 && (base % 4 == 0) &&
          ((reinterpret_cast<uintptr_t>(in) & 15) == 0)) {
        for (int idx = threadIdx.x * 4; idx < nl * N; idx += LB * 4)
          *reinterpret_cast<float4*>(&tile[idx]) =
              *reinterpret_cast<const float4*>(
                  reinterpret_cast<const float*>(in) + base + idx);
      } else {
        for (int idx = threadIdx.x; idx < nl * N; idx += LB)
          tile[idx] = reinterpret_cast<const float*>(in)[base + idx];
      }
    } else {
      for (int idx = threadIdx.x; idx < nl * N; idx += LB)
        tile[idx] = (T)in[l0 * N + idx];
    }
    __syncthreads();
    {
      // compute runs on ALL threads (tail threads chew stale LDS and skip
      // the store): a divergent guard here would block the scalarization of
      // the lane-uniform twiddle loads below
      const T* src = tile + threadIdx.x * N;
      T ar[MCAP], ai[MCAP];
#pragma unroll
      for (int k = 0; k < MCAP; ++k) {
        if (k < m) { ar[k] = T(0); ai[k] = T(0); }
      }
      for (int j = 0; j < N; ++j) {
        const T x = src[j];
        // constant-addrspace cast: lane-uniform twiddles become s_load
        auto twj = (const __attribute__((address_space(4))) T*)
            (tw + (long)j * 2 * m);
#pragma unroll
        for (int k = 0; k < MCAP; ++k) {
          if (k < m) {
            ar[k] += x * twj[2 * k];
            ai[k] += x * twj[2 * k + 1];
          }
        }
      }
      if ((int)threadIdx.x < nl) {   // no continue: barrier above must stay
        T* dst = out + 2 * (l0 + threadIdx.x) * m;   // wave-convergent
#pragma unroll
        for (int k = 0; k < MCAP; ++k) {
          if (k < m) {
            T f = T(1);
            bool edge = (k == 0) || (N % 2 == 0 && 2 * k == N);
            if (factors && !edge) f = T(2);
            dst[2 * k] = f * scale * ar[k];
            dst[2 * k + 1] = (factors && edge) ? T(0) : f * scale * ai[k];
          }
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// C2R (last dim): out[l, j] = scale * Re( sum_k fac_k * in[l, k] * w^{+jk} )
// ---------------------------------------------------------------------------

// Output lines are written through an LDS tile and stored cooperatively
// (coalesced); inputs are 64-128B contiguous per line and read directly.
// acc != nullptr fuses a same-shape addend into the writeback (the block
// residual's input-gradient accumulate: out = idft + acc in one pass instead
// of a separate aten add over the full activation — docs/ROADMAP.md item 4).
// TO != T emits a bf16 output (convert at the writeback; compute stays T).
template <typename T, int MCAP, int NT = 0, int LB = kBlock, typename TO = T>
__global__ __launch_bounds__(LB) void dft_c2r_last_kernel(
    const T* __restrict__ in, TO* __restrict__ out, const T* __restrict__ tw,
    long lines, int N_, int m_, T scale, bool factors,
    const T* __restrict__ acc = nullptr) {
  // NT > 0 pins N and m (== MCAP) at compile time: full unroll + folded
  // twiddle offsets, as in dft_r2c_glds_kernel.  LB: see dft_r2c_last_kernel.
  const int N = NT > 0 ? NT : N_;
  const int m = NT > 0 ? MCAP : m_;
  extern __shared__ __align__(16) char smem_raw[];
  T* tile = reinterpret_cast<T*>(smem_raw);   // [LB * N]

  long ntiles = (lines + LB - 1) / LB;
  for (long tb = blockIdx.x; tb < ntiles; tb += gridDim.x) {
    long l0 = tb * LB;
    int nl = (int)min((long)LB, lines - l0);
    __syncthreads();
    {
      // all threads compute (tail threads re-read line nl-1) so the twiddle
      // loads stay wave-uniform and scalarize; only the LDS/store side is
      // masked via the cooperative writeback bound below
      long lidx = l0 + min((int)threadIdx.x, nl - 1);
      const T* src = in + 2 * lidx * m;
      T yr[MCAP], yi[MCAP];
#pragma unroll
      for (int k = 0; k < MCAP; ++k) {
        if (k < m) {
          T f = scale;
          bool edge = (k == 0) || (N % 2 == 0 && 2 * k == N);
          if (factors && !edge) f = T(2) * scale;
          yr[k] = f * src[2 * k];
          yi[k] = f * src[2 * k + 1];
          if (factors && edge) yi[k] = T(0);
        }
      }
      T* dst = tile + threadIdx.x * N;
      auto syn_j = [&](int j) {
        T sacc = T(0);
        auto twj = (const __attribute__((address_space(4))) T*)
            (tw + (long)(j * 2) * m);          // lane-uniform -> s_load
#pragma unroll
        for (int k = 0; k < MCAP; ++k) {
          if (k < m) {
            // Re(y * w^{+jk}) with w^{+jk} = (twj[2k], twj[2k+1])
            sacc += yr[k] * twj[2 * k] - yi[k] * twj[2 * k + 1];
          }
        }
        dst[j] = sacc;
      };
      if constexpr (NT > 0) {
#pragma unroll
        for (int j = 0; j < NT; ++j) syn_j(j);
      } else {
        for (int j = 0; j < N; ++j) syn_j(j);
      }
    }
    __syncthreads();
    if constexpr (std::is_same<TO, unsigned short>::value) {
      // bf16 output; acc (when set) is a packed-bf16 tensor of the output
      // shape reinterpreted through the fp32-typed parameter — the stash
      // fused residual-grad accumulate for the bf16 model
      const unsigned short* ab = reinterpret_cast<const unsigned short*>(acc);
      const long base = l0 * N;
      if ((nl * N) % 8 == 0 && (base % 8 == 0) &&
          ((reinterpret_cast<uintptr_t>(out) & 15) == 0) &&
          (ab == nullptr ||
           ((reinterpret_cast<uintptr_t>(ab) & 15) == 0))) {
        for (int idx = threadIdx.x * 8; idx < nl * N; idx += LB * 8) {
          float v[8];
#pragma unroll
          for (int e = 0; e < 8; ++e) v[e] = (float)tile[idx + e];
          if (ab != nullptr) {
            const uint4 a8 = *reinterpret_cast<const uint4*>(ab + base + idx);
            const unsigned int w[4] = {a8.x, a8.y, a8.z, a8.w};
#pragma unroll
            for (int q = 0; q < 4; ++q) {
              v[2 * q] += dft_b2f((unsigned short)(w[q] & 0xffffu));
              v[2 * q + 1] += dft_b2f((unsigned short)(w[q] >> 16));
            }
          }
          uint4 raw;
          raw.x = dft_f2b2(v[0], v[1]);
          raw.y = dft_f2b2(v[2], v[3]);
          raw.z = dft_f2b2(v[4], v[5]);
          raw.w = dft_f2b2(v[6], v[7]);
          *reinterpret_cast<uint4*>(out + base + idx) = raw;
        }
      } else {
        for (int idx = threadIdx.x; idx < nl * N; idx += LB) {
          float v = (float)tile[idx];
          if (ab != nullptr) v += dft_b2f(ab[base + idx]);
          __hip_bfloat16 h = __float2bfloat16(v);
          out[base + idx] = *reinterpret_cast<unsigned short*>(&h);
        }
      }
    } else if constexpr (std::is_same<T, float>::value) {
      const long base = l0 * N;
      float* outf = reinterpret_cast<float*>(out);
      if ((nl * N) % 4 == 0 && (base % 4 == 0) &&
          ((reinterpret_cast<uintptr_t>(out) & 15) == 0) &&
          (acc == nullptr || (reinterpret_cast<uintptr_t>(acc) & 15) == 0)) {
        if (acc != nullptr) {
          for (int idx = threadIdx.x * 4; idx < nl * N; idx += LB * 4) {
            const float4 t4 = *reinterpret_cast<const float4*>(&tile[idx]);
            const float4 a4 = *reinterpret_cast<const float4*>(acc + base + idx);
            *reinterpret_cast<float4*>(outf + base + idx) =
                make_float4(t4.x + a4.x, t4.y + a4.y, t4.z + a4.z, t4.w + a4.w);
          }
        } else {
          for (int idx = threadIdx.x * 4; idx < nl * N; idx += LB * 4)
            *reinterpret_cast<float4*>(outf + base + idx) =
                *reinterpret_cast<const float4*>(&tile[idx]);
        }
      } else {
        for (int idx = threadIdx.x; idx < nl * N; idx += LB)
          outf[base + idx] = tile[idx] + (acc ? acc[base + idx] : 0.f);
      }
    } else {
      for (int idx = threadIdx.x; idx < nl * N; idx += LB)
        out[l0 * N + idx] = tile[idx] + (acc ? acc[l0 * N + idx] : T(0));
    }
  }
}

int grid_for_d(long work) {
  long g = (work + kBlock - 1) / kBlock;
  long cap = 256L * 16;
  if (g > cap) g = cap;
  if (g < 1) g = 1;
  return (int)g;
}

void split_dims(const at::Tensor& x, int dim, long& outer, long& inner) {
  outer = 1;
  for (int d = 0; d < dim; ++d) outer *= x.size(d);
  inner = 1;
  for (int d = dim + 1; d < x.dim(); ++d) inner *= x.size(d);
}

#define DFT_MDISPATCH(KERNEL, ...)                                             \
  if (m <= 8) { hipLaunchKernelGGL((KERNEL<scalar_t, 8>), __VA_ARGS__); }      \
  else if (m <= 16) { hipLaunchKernelGGL((KERNEL<scalar_t, 16>), __VA_ARGS__); } \
  else if (m <= 24) { hipLaunchKernelGGL((KERNEL<scalar_t, 24>), __VA_ARGS__); } \
  else { TORCH_CHECK(m <= 32, "dft: m > 32 unsupported natively");             \
         hipLaunchKernelGGL((KERNEL<scalar_t, 32>), __VA_ARGS__); }

// Cached [N, m, 2] twiddle table (fp64 sincos on host, cast to T): one per
// (N, m, direction, dtype) per process; ranks own one device each.
static at::Tensor twiddle_table(int N, int m, bool analysis,
                                const at::TensorOptions& opt) {
  static std::mutex mu;
  static std::unordered_map<long, at::Tensor> cache;
  long key = ((long)N << 20) | ((long)m << 8) | ((long)analysis << 1) |
             (opt.dtype() == at::kFloat ? 1 : 0);
  std::lock_guard<std::mutex> lk(mu);
  auto it = cache.find(key);
  if (it != cache.end()) return it->second;
  double sgn = analysis ? -1.0 : 1.0;
  auto j = at::arange(N, at::kDouble).reshape({N, 1});
  auto k = at::arange(m, at::kDouble).reshape({1, m});
  auto ang = (sgn * 2.0 * M_PI / N) * j * k;
  auto tw = at::stack({ang.cos(), ang.sin()}, 2).to(opt).contiguous();
  cache[key] = tw;
  return tw;
}


}  // namespace

at::Tensor dft_c2c(const at::Tensor& x, int64_t dim, int64_t n,
                   int64_t m_lo, int64_t m_hi, bool analysis, double scale) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous(), "dft_c2c: contiguous GPU input");
  TORCH_CHECK(x.scalar_type() == at::kComplexFloat || x.scalar_type() == at::kComplexDouble,
              "dft_c2c: complex input");
  TORCH_CHECK(n <= kMaxNBig, "dft_c2c: N too large");
  const int m = (int)(m_lo + m_hi);
  TORCH_CHECK(m <= 64 && m <= n, "dft_c2c: bad mode count");
  TORCH_CHECK(x.size(dim) == (analysis ? n : m), "dft_c2c: dim extent mismatch");

  long outer, inner;
  split_dims(x, (int)dim, outer, inner);
  auto sizes = x.sizes().vec();
  sizes[dim] = analysis ? m : n;
  auto out = at::empty(sizes, x.options());
  if (x.numel() == 0) return out;

  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  int grid = grid_for_d(outer * inner);

  bool paired = (m_hi == m_lo) && (inner % 2 == 0) && m_lo <= 16;
  static const bool no_radix = []() {
    const char* e = getenv("DFNO_DFT_NO_RADIX");  // A/B knob
    return e && e[0] == '1';
  }();
  // j-split degree: raise wave occupancy when outer*pairs alone underfills
  // the 1024 SIMDs (x/y-dim transforms); G must divide n for the chain wrap
  int Gsel = 1;
  if (paired) {
    static const long gtarget = []() {
      const char* e = getenv("DFNO_DFT_GTARGET");  // threads wanted (tunable)
      // default 1 = j-split off: measured on MI355X, G>1 LOSES at every
      // flagship shape (y-dim c2c 0.119 -> 0.137 ms at G=4) -- the per-wave
      // load stream already saturates at 0.5 waves/SIMD and narrowing the
      // contiguous chunk from 1 KB to 256 B costs more than the extra
      // occupancy buys.  Kept as a knob for other shapes.
      return e ? atol(e) : 1L;
    }();
    long pwork = outer * (inner / 2);
    while (Gsel < 8 && (n % (Gsel * 2) == 0) && pwork * Gsel < gtarget)
      Gsel *= 2;
  }
  int grid2 = paired ? grid_for_d(outer * (inner / 2) * Gsel) : grid;
#define DFT_LG(KERNEL, LC, ...)                                                \
  switch (Gsel) {                                                              \
    case 8: hipLaunchKernelGGL((KERNEL<scalar_t, LC, 8>), __VA_ARGS__); break; \
    case 4: hipLaunchKernelGGL((KERNEL<scalar_t, LC, 4>), __VA_ARGS__); break; \
    case 2: hipLaunchKernelGGL((KERNEL<scalar_t, LC, 2>), __VA_ARGS__); break; \
    default: hipLaunchKernelGGL((KERNEL<scalar_t, LC, 1>), __VA_ARGS__);       \
  }
#define DFT_LDISPATCH(KERNEL, ...)                                             \
  if (m_lo <= 8) { DFT_LG(KERNEL, 9, __VA_ARGS__) }                            \
  else if (m_lo <= 12) { DFT_LG(KERNEL, 13, __VA_ARGS__) }                     \
  else { DFT_LG(KERNEL, 17, __VA_ARGS__) }
  at::Tensor tw;
  if (paired)
    tw = twiddle_table((int)n, (int)m_lo + 1, analysis,
                       x.options().dtype(c10::toRealValueType(x.scalar_type())));
  // radix-8xNB eligibility: N in {64, 128, 256} (NBT 8/16/32), fp32, paired
  bool radix_ok = paired && (n == 64 || n == 128 || n == 256) &&
                  c10::toRealValueType(x.scalar_type()) == at::kFloat &&
                  !no_radix;
#define DFT_RADIX_LC(KERNEL, NBT)                                              \
  if (n == 64 && m_lo == 12) { /* flagship: fully folded */                    \
    hipLaunchKernelGGL((KERNEL<scalar_t, 13, 12, 8>), dim3(grid2),             \
                       dim3(kBlock), 0, stream, inp, op,                       \
                       tw.data_ptr<scalar_t>(), outer, inner, (int)m_lo,       \
                       (scalar_t)scale);                                       \
  } else if (m_lo <= 8) {                                                      \
    hipLaunchKernelGGL((KERNEL<scalar_t, 9, 0, NBT>), dim3(grid2),             \
                       dim3(kBlock), 0, stream, inp, op,                       \
                       tw.data_ptr<scalar_t>(), outer, inner, (int)m_lo,       \
                       (scalar_t)scale);                                       \
  } else if (m_lo <= 12) {                                                     \
    hipLaunchKernelGGL((KERNEL<scalar_t, 13, 0, NBT>), dim3(grid2),            \
                       dim3(kBlock), 0, stream, inp, op,                       \
                       tw.data_ptr<scalar_t>(), outer, inner, (int)m_lo,       \
                       (scalar_t)scale);                                       \
  } else {                                                                     \
    hipLaunchKernelGGL((KERNEL<scalar_t, 17, 0, NBT>), dim3(grid2),            \
                       dim3(kBlock), 0, stream, inp, op,                       \
                       tw.data_ptr<scalar_t>(), outer, inner, (int)m_lo,       \
                       (scalar_t)scale);                                       \
  }
#define DFT_RADIX(KERNEL)                                                      \
  if (n == 256) { DFT_RADIX_LC(KERNEL, 32) }                                   \
  else if (n == 128) { DFT_RADIX_LC(KERNEL, 16) }                              \
  else { DFT_RADIX_LC(KERNEL, 8) }
  AT_DISPATCH_FLOATING_TYPES(c10::toRealValueType(x.scalar_type()), "dft_c2c", [&] {
    auto inp = reinterpret_cast<const scalar_t*>(x.data_ptr());
    auto op = reinterpret_cast<scalar_t*>(out.data_ptr());
    if (radix_ok && analysis) {
      DFT_RADIX(dft_c2c_radix8_ana_kernel)
    } else if (paired && analysis) {
      DFT_LDISPATCH(dft_c2c_analysis2_kernel, dim3(grid2), dim3(kBlock), 0,
                    stream, inp, op, tw.data_ptr<scalar_t>(), outer, (int)n,
                    inner, (int)m_lo, (scalar_t)scale)
    } else if (radix_ok && !analysis) {
      DFT_RADIX(dft_c2c_radix8_syn_kernel)
    } else if (paired) {
      DFT_LDISPATCH(dft_c2c_synthesis2_kernel, dim3(grid2), dim3(kBlock), 0,
                    stream, inp, op, tw.data_ptr<scalar_t>(), outer, (int)n,
                    inner, (int)m_lo, (scalar_t)scale)
    } else if (analysis) {
      DFT_MDISPATCH(dft_c2c_analysis_kernel, dim3(grid), dim3(kBlock), 0, stream,
                    inp, op, outer, (int)n, inner, (int)m_lo, (int)m_hi,
                    (scalar_t)scale)
    } else {
      DFT_MDISPATCH(dft_c2c_synthesis_kernel, dim3(grid), dim3(kBlock), 0, stream,
                    inp, op, outer, (int)n, inner, (int)m_lo, (int)m_hi,
                    (scalar_t)scale)
    }
  });
#undef DFT_RADIX
#undef DFT_RADIX_LC
#undef DFT_LDISPATCH
#undef DFT_LG
  DFNO_CHECK_LAUNCH("dft_c2c");
  return out;
}

static at::Tensor dft_r2c_impl(const at::Tensor& x, int64_t dim, int64_t m,
                               double scale, bool factors) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous(), "dft_r2c: contiguous GPU input");
  TORCH_CHECK(dim == x.dim() - 1, "dft_r2c: last-dim only");
  const int N = (int)x.size(dim);
  TORCH_CHECK(N <= kMaxNBig && m <= N, "dft_r2c: bad sizes");
  TORCH_CHECK(m <= 32, "dft_r2c: m > 32 unsupported natively");
  long lines = x.numel() / std::max(N, 1);

  const bool bf16_in = x.scalar_type() == at::kBFloat16;
  auto sizes = x.sizes().vec();
  sizes[dim] = m;
  auto out = at::empty(sizes, x.options().dtype(
      x.scalar_type() == at::kDouble ? at::kComplexDouble : at::kComplexFloat));
  if (x.numel() == 0) return out;

  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  long ntiles = (lines + kBlock - 1) / kBlock;
  int grid = (int)std::min(ntiles, 4096L);
  auto tw = twiddle_table(N, (int)m, /*analysis=*/true,
                          x.options().dtype(bf16_in ? at::kFloat
                                                    : x.scalar_type()));
  if (bf16_in) {
    // bf16-IO variant: bf16 staged + converted at the LDS write, fp32
    // compute, complex64 out — the bf16 model's real-side transforms read
    // half the bytes and skip the boundary-cast pass entirely
    auto inp = reinterpret_cast<const unsigned short*>(x.data_ptr());
    auto op = reinterpret_cast<float*>(out.data_ptr());
    bool glds_ok = N <= 64 &&
                   ((reinterpret_cast<uintptr_t>(inp) & 15) == 0) &&
                   ((long)kBlock * N) % 8 == 0 && lines * N >= 8;
    bool mexact = (m == 8 || m == 16 || m == 24 || m == 32);
    if (glds_ok && mexact && (N == 30 || N == 64 || N == 40 || N == 32)) {
      // DMA the raw bf16 tile (glds ring at half the fp32 footprint)
      size_t smem2 = 2 * sizeof(unsigned short) * (size_t)kBlock * N;
#define R2CGB(MC, NGV, NTV)                                                    \
      hipLaunchKernelGGL((dft_r2c_glds_kernel<float, MC, NGV, NTV,             \
                                              unsigned short>),                \
                         dim3(grid), dim3(kBlock), smem2, stream, inp, op,     \
                         tw.data_ptr<float>(), lines, N, (int)m,               \
                         (float)scale, factors);
#define R2CGB_M(NGV, NTV)                                                      \
      if (m == 8) { R2CGB(8, NGV, NTV) } else if (m == 16) { R2CGB(16, NGV, NTV) } \
      else if (m == 24) { R2CGB(24, NGV, NTV) } else { R2CGB(32, NGV, NTV) }
      if (N == 30) { R2CGB_M(4, 30) }
      else if (N == 64) { R2CGB_M(8, 64) }
      else if (N == 40) { R2CGB_M(8, 40) }
      else { R2CGB_M(4, 32) }
#undef R2CGB_M
#undef R2CGB
      DFNO_CHECK_LAUNCH("dft_r2c_bf16");
      return out;
    }
#define R2C_BF(MC, LBV)                                                        \
    { long nt2 = (lines + LBV - 1) / LBV;                                      \
      int grid2 = (int)std::min(nt2, 2048L);                                   \
      size_t smem2 = sizeof(float) * (size_t)LBV * N;                          \
      hipLaunchKernelGGL((dft_r2c_last_kernel<float, MC, LBV, unsigned short>),\
                         dim3(grid2), dim3(LBV), smem2, stream, inp, op,       \
                         tw.data_ptr<float>(), lines, N, (int)m,               \
                         (float)scale, factors); }
#define R2C_BF_M(LBV)                                                          \
    if (m <= 8) { R2C_BF(8, LBV) } else if (m <= 16) { R2C_BF(16, LBV) }       \
    else if (m <= 24) { R2C_BF(24, LBV) } else { R2C_BF(32, LBV) }
    if (N <= 64) { R2C_BF_M(256) }
    else if (N <= 128) { R2C_BF_M(128) }
    else { R2C_BF_M(64) }
#undef R2C_BF_M
#undef R2C_BF
    DFNO_CHECK_LAUNCH("dft_r2c_bf16");
    return out;
  }
#define R2CG(MC, NGV, NTV)                                                     \
      hipLaunchKernelGGL((dft_r2c_glds_kernel<scalar_t, MC, NGV, NTV>),        \
                         dim3(grid), dim3(kBlock), smem2, stream, inp, op,     \
                         tw.data_ptr<scalar_t>(), lines, N, (int)m,            \
                         (scalar_t)scale, factors);
#define R2CG_M(NGV, NTV)                                                       \
      if (m <= 8) { R2CG(8, NGV, NTV) } else if (m <= 16) { R2CG(16, NGV, NTV) } \
      else if (m <= 24) { R2CG(24, NGV, NTV) } else { R2CG(32, NGV, NTV) }
#define R2C_BIG(MC, LBV)                                                       \
      { long nt2 = (lines + LBV - 1) / LBV;                                    \
        int grid2 = (int)std::min(nt2, 2048L);                                 \
        size_t smem2 = sizeof(scalar_t) * (size_t)LBV * N;                     \
        hipLaunchKernelGGL((dft_r2c_last_kernel<scalar_t, MC, LBV>),           \
                           dim3(grid2), dim3(LBV), smem2, stream, inp, op,     \
                           tw.data_ptr<scalar_t>(), lines, N, (int)m,          \
                           (scalar_t)scale, factors); }
#define R2C_BIG_M(LBV)                                                         \
      if (m <= 8) { R2C_BIG(8, LBV) } else if (m <= 16) { R2C_BIG(16, LBV) }   \
      else if (m <= 24) { R2C_BIG(24, LBV) } else { R2C_BIG(32, LBV) }
  AT_DISPATCH_FLOATING_TYPES(x.scalar_type(), "dft_r2c", [&] {
    size_t smem = sizeof(scalar_t) * (size_t)kBlock * N;
    auto inp = x.data_ptr<scalar_t>();
    auto op = reinterpret_cast<scalar_t*>(out.data_ptr());
    if (N > kMaxN) {
      // big-N path: smaller lines-per-tile keeps the LDS tile <= 64 KiB
      // (2 blocks/CU); one thread per line, table-driven naive DFT
      if (N <= 128) { R2C_BIG_M(128) } else { R2C_BIG_M(64) }
      return;
    }
    bool glds_ok = std::is_same<scalar_t, float>::value &&
                   ((reinterpret_cast<uintptr_t>(inp) & 15) == 0) &&
                   ((long)kBlock * N) % 4 == 0 && lines * N >= 4;
    if (glds_ok) {
      size_t smem2 = 2 * smem;   // double-buffered ring
      // common N with m on an MCAP boundary get the fully-folded fast path
      bool mexact = (m == 8 || m == 16 || m == 24 || m == 32);
      if (mexact && N == 30) { R2CG_M(8, 30) }
      else if (mexact && N == 64) { R2CG_M(16, 64) }
      else if (mexact && N == 40) { R2CG_M(16, 40) }
      else if (mexact && N == 32) { R2CG_M(8, 32) }
      else if (N <= 32) { R2CG_M(8, 0) }   // ceil(N/4) <= 8 glds per wave
      else { R2CG_M(16, 0) }
    } else {
      DFT_MDISPATCH(dft_r2c_last_kernel, dim3(grid), dim3(kBlock), smem,
                    stream, inp, op, tw.data_ptr<scalar_t>(), lines, N,
                    (int)m, (scalar_t)scale, factors)
    }
  });
#undef R2C_BIG_M
#undef R2C_BIG
#undef R2CG_M
#undef R2CG
  DFNO_CHECK_LAUNCH("dft_r2c");
  return out;
}

static at::Tensor dft_c2r_impl(const at::Tensor& y, int64_t dim, int64_t n_out,
                               double scale, bool factors,
                               const at::Tensor& accum = at::Tensor(),
                               bool out_bf16 = false) {
  TORCH_CHECK(y.is_cuda() && y.is_contiguous(), "dft_c2r: contiguous GPU input");
  TORCH_CHECK(dim == y.dim() - 1, "dft_c2r: last-dim only");
  TORCH_CHECK(y.scalar_type() == at::kComplexFloat || y.scalar_type() == at::kComplexDouble,
              "dft_c2r: complex input");
  const int m = (int)y.size(dim);
  const int N = (int)n_out;
  TORCH_CHECK(N <= kMaxNBig && m <= N, "dft_c2r: bad sizes");
  TORCH_CHECK(m <= 32, "dft_c2r: m > 32 unsupported natively");
  long lines = y.numel() / std::max(m, 1);

  auto sizes = y.sizes().vec();
  sizes[dim] = N;
  if (out_bf16) {
    TORCH_CHECK(y.scalar_type() == at::kComplexFloat,
                "dft_c2r: bf16 output needs c64 input");
    TORCH_CHECK(!accum.defined() || accum.numel() == 0 ||
                (accum.scalar_type() == at::kBFloat16 &&
                 accum.is_contiguous()),
                "dft_c2r: bf16 accumulate must be contiguous bf16");
  }
  auto out = at::empty(sizes, y.options().dtype(
      out_bf16 ? at::kBFloat16
               : (y.scalar_type() == at::kComplexFloat ? at::kFloat
                                                       : at::kDouble)));
  if (y.numel() == 0) return out;

  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  long ntiles = (lines + kBlock - 1) / kBlock;
  int grid = (int)std::min(ntiles, 4096L);
  auto tw = twiddle_table(N, m, /*analysis=*/false,
                          out.options().dtype(out_bf16 ? at::kFloat
                                                       : out.scalar_type()));
  if (out_bf16) {
    // bf16-output writeback (fp32 compute/LDS tile), flagship NT folding kept
    auto inp = reinterpret_cast<const float*>(y.data_ptr());
    auto op = reinterpret_cast<unsigned short*>(out.data_ptr());
    const float* accb = nullptr;               // packed bf16 through T*
    if (accum.defined() && accum.numel() > 0) {
      TORCH_CHECK(accum.numel() == out.numel(), "dft_c2r: accumulate shape");
      accb = reinterpret_cast<const float*>(accum.data_ptr());
    }
#define C2RB(MC, NTV, LBV)                                                     \
    { long nt2 = (lines + LBV - 1) / LBV;                                      \
      int grid2 = (int)std::min(nt2, LBV == kBlock ? 4096L : 2048L);           \
      size_t smem2 = sizeof(float) * (size_t)LBV * N;                          \
      hipLaunchKernelGGL(                                                      \
          (dft_c2r_last_kernel<float, MC, NTV, LBV, unsigned short>),          \
          dim3(grid2), dim3(LBV), smem2, stream, inp, op,                      \
          tw.data_ptr<float>(), lines, N, (int)m, (float)scale, factors,       \
          accb); }
#define C2RB_M(NTV, LBV)                                                       \
    if (m <= 8) { C2RB(8, NTV, LBV) } else if (m <= 16) { C2RB(16, NTV, LBV) } \
    else if (m <= 24) { C2RB(24, NTV, LBV) } else { C2RB(32, NTV, LBV) }
    const bool mexact = (m == 8 || m == 16 || m == 24 || m == 32);
    if (mexact && N == 30) { C2RB_M(30, kBlock) }
    else if (mexact && N == 64) { C2RB_M(64, kBlock) }
    else if (N <= 64) { C2RB_M(0, kBlock) }
    else if (N <= 128) { C2RB_M(0, 128) }
    else { C2RB_M(0, 64) }
#undef C2RB_M
#undef C2RB
    DFNO_CHECK_LAUNCH("dft_c2r_bf16");
    return out;
  }
#define C2RG(MC, NTV)                                                          \
      hipLaunchKernelGGL((dft_c2r_last_kernel<scalar_t, MC, NTV>),             \
                         dim3(grid), dim3(kBlock), smem, stream, inp, op,      \
                         tw.data_ptr<scalar_t>(), lines, N, (int)m,            \
                         (scalar_t)scale, factors, accp);
#define C2R_BIG(MC, LBV)                                                       \
      { long nt2 = (lines + LBV - 1) / LBV;                                    \
        int grid2 = (int)std::min(nt2, 2048L);                                 \
        size_t smem2 = sizeof(scalar_t) * (size_t)LBV * N;                     \
        hipLaunchKernelGGL((dft_c2r_last_kernel<scalar_t, MC, 0, LBV>),        \
                           dim3(grid2), dim3(LBV), smem2, stream, inp, op,     \
                           tw.data_ptr<scalar_t>(), lines, N, (int)m,          \
                           (scalar_t)scale, factors, accp); }
#define C2R_BIG_M(LBV)                                                         \
      if (m <= 8) { C2R_BIG(8, LBV) } else if (m <= 16) { C2R_BIG(16, LBV) }   \
      else if (m <= 24) { C2R_BIG(24, LBV) } else { C2R_BIG(32, LBV) }
  if (accum.defined() && accum.numel() > 0) {
    TORCH_CHECK(accum.is_cuda() && accum.is_contiguous() &&
                accum.numel() == out.numel() &&
                accum.scalar_type() == out.scalar_type(),
                "dft_c2r: bad accumulate tensor");
  }
  AT_DISPATCH_FLOATING_TYPES(out.scalar_type(), "dft_c2r", [&] {
    size_t smem = sizeof(scalar_t) * (size_t)kBlock * N;
    auto inp = reinterpret_cast<const scalar_t*>(y.data_ptr());
    auto op = out.data_ptr<scalar_t>();
    const scalar_t* accp = (accum.defined() && accum.numel() > 0)
                               ? accum.data_ptr<scalar_t>() : nullptr;
    if (N > kMaxN) {
      if (N <= 128) { C2R_BIG_M(128) } else { C2R_BIG_M(64) }
      return;
    }
    bool mexact = std::is_same<scalar_t, float>::value &&
                  (m == 8 || m == 16 || m == 24 || m == 32);
    if (mexact && N == 30) {
      if (m == 8) { C2RG(8, 30) } else if (m == 16) { C2RG(16, 30) }
      else if (m == 24) { C2RG(24, 30) } else { C2RG(32, 30) }
    } else if (mexact && N == 64) {
      if (m == 8) { C2RG(8, 64) } else if (m == 16) { C2RG(16, 64) }
      else if (m == 24) { C2RG(24, 64) } else { C2RG(32, 64) }
    } else if (mexact && N == 40) {
      if (m == 8) { C2RG(8, 40) } else if (m == 16) { C2RG(16, 40) }
      else if (m == 24) { C2RG(24, 40) } else { C2RG(32, 40) }
    } else {
      DFT_MDISPATCH(dft_c2r_last_kernel, dim3(grid), dim3(kBlock), smem,
                    stream, inp, op, tw.data_ptr<scalar_t>(), lines, N,
                    (int)m, (scalar_t)scale, factors, accp)
    }
  });
#undef C2R_BIG_M
#undef C2R_BIG
#undef C2RG
  DFNO_CHECK_LAUNCH("dft_c2r");
  return out;
}

at::Tensor dft_rfft_trunc(const at::Tensor& x, int64_t dim, int64_t m) {
  return dft_r2c_impl(x, dim, m, 1.0, /*factors=*/false);
}

at::Tensor dft_rfft_trunc_adj(const at::Tensor& gy, int64_t dim, int64_t n) {
  // gx_j = Re(sum_k gY_k w^{+jk}) — c2r with unit factors / unit scale
  return dft_c2r_impl(gy, dim, n, 1.0, /*factors=*/false);
}

at::Tensor dft_rfft_trunc_adj_acc(const at::Tensor& gy, int64_t dim, int64_t n,
                                  const at::Tensor& accum) {
  // adjoint with a fused addend: gx = idft(gy) + accum in one writeback
  // (the block residual's input-grad accumulate)
  return dft_c2r_impl(gy, dim, n, 1.0, /*factors=*/false, accum);
}

at::Tensor dft_pad_irfft(const at::Tensor& y, int64_t dim, int64_t n_out, int64_t m) {
  TORCH_CHECK(y.size(dim) == m, "dft_pad_irfft: m mismatch");
  return dft_c2r_impl(y, dim, n_out, 1.0 / (double)n_out, /*factors=*/true);
}

at::Tensor dft_pad_irfft_bf16(const at::Tensor& y, int64_t dim, int64_t n_out,
                              int64_t m) {
  // bf16-output variant for the bf16 compute config (no boundary cast)
  TORCH_CHECK(y.size(dim) == m, "dft_pad_irfft: m mismatch");
  return dft_c2r_impl(y, dim, n_out, 1.0 / (double)n_out, /*factors=*/true,
                      at::Tensor(), /*out_bf16=*/true);
}

at::Tensor dft_rfft_trunc_adj_bf16(const at::Tensor& gy, int64_t dim,
                                   int64_t n, const at::Tensor& accum) {
  // bf16-output adjoint (the bf16 model's rfft input gradient); accum is
  // the optional stashed residual gradient fused into the writeback
  return dft_c2r_impl(gy, dim, n, 1.0, /*factors=*/false, accum,
                      /*out_bf16=*/true);
}

at::Tensor dft_pad_irfft_adj(const at::Tensor& gx, int64_t dim, int64_t m) {
  const int N = (int)gx.size(dim);
  return dft_r2c_impl(gx, dim, m, 1.0 / (double)N, /*factors=*/true);
}
