// Bisection probe for the staged-kernel throughput wall: replicate the
// dft_r2c_last skeleton (stage 256 lines x N floats -> barrier -> per-thread
// line compute -> coalesced write) and toggle each element.
#include <hip/hip_runtime.h>
#include <cstdio>

constexpr int KB = 256;

// MODE: 0 = full (stage+compute+write)
//       1 = no compute (stage + trivial write)
//       2 = no stage (compute reads global directly, strided per thread)
//       3 = stage + compute, no barrier (WRONG results; timing only)
//       4 = stage + write, compute replaced by 8-iter dummy (cheap compute)
template <int N, int M, int MODE>
__global__ __launch_bounds__(KB) void r2clike(const float* __restrict__ in,
                                              float* __restrict__ out,
                                              const float* __restrict__ tw,
                                              long lines) {
  __shared__ float tile[KB * N];
  long ntiles = (lines + KB - 1) / KB;
  for (long tb = blockIdx.x; tb < ntiles; tb += gridDim.x) {
    long l0 = tb * KB;
    if constexpr (MODE != 2) {
      if constexpr (MODE != 3) __syncthreads();
      const long base = l0 * N;
      for (int idx = threadIdx.x * 4; idx < KB * N; idx += KB * 4)
        *reinterpret_cast<float4*>(&tile[idx]) =
            *reinterpret_cast<const float4*>(in + base + idx);
      if constexpr (MODE != 3) __syncthreads();
    }
    float ar[M], ai[M];
#pragma unroll
    for (int k = 0; k < M; ++k) { ar[k] = ai[k] = 0.f; }
    if constexpr (MODE == 0 || MODE == 3) {
      const float* src = tile + threadIdx.x * N;
      for (int j = 0; j < N; ++j) {
        const float x = src[j];
        auto twj = (const __attribute__((address_space(4))) float*)(tw + j * 2 * M);
#pragma unroll
        for (int k = 0; k < M; ++k) {
          ar[k] += x * twj[2 * k];
          ai[k] += x * twj[2 * k + 1];
        }
      }
    } else if constexpr (MODE == 2) {
      const float* src = in + (l0 + threadIdx.x) * N;
      for (int j = 0; j < N; ++j) {
        const float x = src[j];
        auto twj = (const __attribute__((address_space(4))) float*)(tw + j * 2 * M);
#pragma unroll
        for (int k = 0; k < M; ++k) {
          ar[k] += x * twj[2 * k];
          ai[k] += x * twj[2 * k + 1];
        }
      }
    } else if constexpr (MODE == 4) {
      const float* src = tile + threadIdx.x * N;
#pragma unroll
      for (int k = 0; k < M; ++k) { ar[k] = src[k]; ai[k] = src[k + M]; }
    } else {  // MODE 1
      const float* src = tile + threadIdx.x * N;
#pragma unroll
      for (int k = 0; k < M; ++k) { ar[k] = src[k]; ai[k] = src[k + M]; }
    }
    float* dst = out + 2 * (l0 + threadIdx.x) * M;
#pragma unroll
    for (int k = 0; k < M; ++k) { dst[2 * k] = ar[k]; dst[2 * k + 1] = ai[k]; }
  }
}

// MODE 5: full + per-block phase skew (s_sleep staggers co-resident blocks
// so their stage/compute phases anti-align -> tests the convoy hypothesis)
template <int N, int M>
__global__ __launch_bounds__(KB) void r2cskew(const float* __restrict__ in,
                                              float* __restrict__ out,
                                              const float* __restrict__ tw,
                                              long lines) {
  __shared__ float tile[KB * N];
  // ~450-cycle steps: five co-resident blocks spread across one tile period
  for (int z = 0; z < (int)(blockIdx.x % 5); ++z)
    __builtin_amdgcn_s_sleep(7);
  long ntiles = (lines + KB - 1) / KB;
  for (long tb = blockIdx.x; tb < ntiles; tb += gridDim.x) {
    long l0 = tb * KB;
    __syncthreads();
    const long base = l0 * N;
    for (int idx = threadIdx.x * 4; idx < KB * N; idx += KB * 4)
      *reinterpret_cast<float4*>(&tile[idx]) =
          *reinterpret_cast<const float4*>(in + base + idx);
    __syncthreads();
    float ar[M], ai[M];
#pragma unroll
    for (int k = 0; k < M; ++k) { ar[k] = ai[k] = 0.f; }
    const float* src = tile + threadIdx.x * N;
    for (int j = 0; j < N; ++j) {
      const float x = src[j];
      auto twj = (const __attribute__((address_space(4))) float*)(tw + j * 2 * M);
#pragma unroll
      for (int k = 0; k < M; ++k) {
        ar[k] += x * twj[2 * k];
        ai[k] += x * twj[2 * k + 1];
      }
    }
    float* dst = out + 2 * (l0 + threadIdx.x) * M;
#pragma unroll
    for (int k = 0; k < M; ++k) { dst[2 * k] = ar[k]; dst[2 * k + 1] = ai[k]; }
  }
}

// MODE 6: glds double-buffer ring — stage tile t+1 via global_load_lds while
// computing tile t (counted vmcnt, raw barrier).  True intra-block overlap.
template <int N, int M>
__global__ __launch_bounds__(KB) void r2cglds(const float* __restrict__ in,
                                              float* __restrict__ out,
                                              const float* __restrict__ tw,
                                              long lines) {
  constexpr int TILEF = KB * N;              // floats per tile (7680)
  constexpr int NG = (TILEF / 4 + KB - 1) / KB;  // glds per wave (pad) = 8
  __shared__ float ring[2][TILEF];
  const int wave = threadIdx.x / 64, lane = threadIdx.x % 64;
  long ntiles = (lines + KB - 1) / KB;

  auto issue = [&](int buf, long tb) {
    const float* src = in + tb * KB * N;
    float* dst = ring[buf];
#pragma unroll
    for (int k = 0; k < NG; ++k) {
      int slot = wave * 64 + k * KB + lane;
      int fo = slot * 4;
      if (fo >= TILEF) fo = TILEF - 4;       // pad: clamp
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)(src + fo),
          (__attribute__((address_space(3))) void*)&dst[fo], 16, 0, 0);
    }
  };

  long nt_mine = 0;
  long first = blockIdx.x;
  if (first < ntiles) issue(0, first);
  long c = 0;
  for (long tb = first; tb < ntiles; tb += gridDim.x, ++c) {
    long nxt = tb + gridDim.x;
    if (nxt < ntiles) {
      issue((int)((c + 1) & 1), nxt);
      asm volatile("s_waitcnt vmcnt(%0)" ::"n"(NG) : "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();
    asm volatile("" ::: "memory");
    const float* tile = ring[c & 1];
    float ar[M], ai[M];
#pragma unroll
    for (int k = 0; k < M; ++k) { ar[k] = ai[k] = 0.f; }
    const float* src = tile + threadIdx.x * N;
    for (int j = 0; j < N; ++j) {
      const float x = src[j];
      auto twj = (const __attribute__((address_space(4))) float*)(tw + j * 2 * M);
#pragma unroll
      for (int k = 0; k < M; ++k) {
        ar[k] += x * twj[2 * k];
        ai[k] += x * twj[2 * k + 1];
      }
    }
    long l0 = tb * KB;
    float* dst = out + 2 * (l0 + threadIdx.x) * M;
#pragma unroll
    for (int k = 0; k < M; ++k) { dst[2 * k] = ar[k]; dst[2 * k + 1] = ai[k]; }
    // all waves done reading ring[c&1] before its re-issue 2 iters later:
    // one barrier per iteration gives that spacing with 2 buffers + 1-deep
    // prefetch ONLY if re-issue targets the buffer read LAST iter -> need a
    // second barrier before issue; cheaper: barrier here at loop end
    __builtin_amdgcn_s_barrier();
    asm volatile("" ::: "memory");
  }
  (void)nt_mine;
}

int main() {
  constexpr int N = 30, M = 8;
  long lines = 5242880;
  float *in, *out, *tw;
  hipMalloc(&in, lines * N * 4);
  hipMalloc(&out, lines * 2 * M * 4);
  hipMalloc(&tw, N * 2 * M * 4);
  hipMemset(in, 1, lines * N * 4);
  hipMemset(tw, 1, N * 2 * M * 4);
  double gb = (lines * N * 4 + lines * 2 * M * 4) / 1e9;
  auto run = [&](const char* name, auto k, int grid) {
    hipLaunchKernelGGL(k, dim3(grid), dim3(KB), 0, 0, in, out, tw, lines);
    hipDeviceSynchronize();
    hipEvent_t a, b; hipEventCreate(&a); hipEventCreate(&b);
    hipEventRecord(a);
    for (int r = 0; r < 5; ++r)
      hipLaunchKernelGGL(k, dim3(grid), dim3(KB), 0, 0, in, out, tw, lines);
    hipEventRecord(b); hipEventSynchronize(b);
    float ms; hipEventElapsedTime(&ms, a, b); ms /= 5;
    printf("%-34s %7.3f ms  %5.2f TB/s\n", name, ms, gb / ms * 1000 / 1000);
  };
  for (int grid : {4096, 20480}) {
    printf("-- grid %d --\n", grid);
    run("full (stage+compute+write)", r2clike<N, M, 0>, grid);
    run("no-compute", r2clike<N, M, 1>, grid);
    run("no-stage (direct strided reads)", r2clike<N, M, 2>, grid);
    run("no-barrier (timing only)", r2clike<N, M, 3>, grid);
    run("skewed phases", r2cskew<N, M>, grid);
    run("glds 2-ring", r2cglds<N, M>, grid);
  }
  return 0;
}
