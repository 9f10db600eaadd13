// Streaming-BW probe: read-sum 1.3 GB with varying register ballast
// (occupancy) and access granularity. Answers: what occupancy does HBM3E
// need, and does 64B-strided access really cost nothing?
#include <hip/hip_runtime.h>
#include <cstdio>
#include <vector>

template <int BALLAST, int STRIDED>
__global__ __launch_bounds__(256) void readsum(const float4* __restrict__ in,
                                               float* __restrict__ out, long n4) {
  float b[BALLAST];
#pragma unroll
  for (int i = 0; i < BALLAST; ++i) b[i] = (float)threadIdx.x + i;
  float acc = 0.f;
  long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  if constexpr (STRIDED) {
    // 64B chunks, 4 lanes together, chunks 1KB apart (z-dim c2c pattern)
    long nchunk = n4 / 4;
    for (long c = i0; c < nchunk; c += stride) {
      long chunk = c / 4, lane4 = c % 4;
      long base = (chunk * 16 + lane4) % n4;   // scatter chunks
      float4 v = in[base];
      acc += v.x + v.y + v.z + v.w;
    }
  } else {
    for (long i = i0; i < n4; i += stride) {
      float4 v = in[i];
      acc += v.x + v.y + v.z + v.w;
    }
  }
#pragma unroll
  for (int i = 0; i < BALLAST; ++i) acc += b[i] * 1e-30f;
  if (acc == 12345.678f) out[0] = acc;  // never true; keeps acc alive
}

int main() {
  long n4 = 320L * 1024 * 1024 / 4;  // 1.34 GB of float4
  float4* d; hipMalloc(&d, n4 * sizeof(float4) / 4 * 4);
  hipMalloc(&d, n4 * 16);
  float* o; hipMalloc(&o, 4);
  hipMemset(d, 1, n4 * 16);
  auto run = [&](const char* name, auto kern, int grid) {
    hipLaunchKernelGGL(kern, dim3(grid), dim3(256), 0, 0, d, o, n4);
    hipDeviceSynchronize();
    hipEvent_t a, b; hipEventCreate(&a); hipEventCreate(&b);
    hipEventRecord(a);
    for (int r = 0; r < 5; ++r)
      hipLaunchKernelGGL(kern, dim3(grid), dim3(256), 0, 0, d, o, n4);
    hipEventRecord(b); hipEventSynchronize(b);
    float ms; hipEventElapsedTime(&ms, a, b); ms /= 5;
    printf("%-28s grid=%5d  %7.3f ms  %6.2f TB/s\n", name, grid, ms,
           n4 * 16.0 / (ms * 1e-3) / 1e12);
  };
  for (int grid : {1024, 4096, 16384}) {
    run("ballast8   (8+ waves)", readsum<8, 0>, grid);
    run("ballast120 (4 waves)", readsum<120, 0>, grid);
    run("ballast180 (2 waves)", readsum<180, 0>, grid);
    run("ballast240 (1-2 waves)", readsum<240, 0>, grid);
  }
  run("strided64B ballast8", readsum<8, 1>, 4096);
  run("strided64B ballast180", readsum<180, 1>, 4096);
  return 0;
}
