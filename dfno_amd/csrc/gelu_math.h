// Shared exact-GELU device math.
//
// float: Abramowitz-Stegun 7.1.26 rational erf (|err| <= 1.5e-7, well inside
// the fp32 tolerances used against torch's F.gelu) with the exp(-z^2/2)
// factor SHARED between gelu and its derivative — ocml erff alone is ~42
// VALU + 2 branches, so the fused value+grad path is ~2x cheaper.
// double: ocml erf/exp (kept bit-accurate for the fp64 test paths).
#pragma once

#include <hip/hip_runtime.h>

namespace dfno_gelu {

constexpr double kInvSqrt2 = 0.7071067811865476;
constexpr double kInvSqrt2Pi = 0.3989422804014327;

// erf(|x|) from E = exp(-x*x); A&S 7.1.26
__device__ __forceinline__ float erf_from_E(float ax, float E) {
  float t = 1.0f / (1.0f + 0.3275911f * ax);
  float p = t * (0.254829592f +
            t * (-0.284496736f +
            t * (1.421413741f +
            t * (-1.453152027f + t * 1.061405429f))));
  return 1.0f - p * E;
}

// gelu(z) = 0.5 z (1 + erf(z/sqrt2))
__device__ __forceinline__ float gelu(float z) {
  float x = z * (float)kInvSqrt2;
  float E = __expf(-x * x);
  float e = erf_from_E(fabsf(x), E);
  e = copysignf(e, x);
  return 0.5f * z * (1.0f + e);
}

__device__ __forceinline__ float gelu_grad(float z) {
  float x = z * (float)kInvSqrt2;
  float E = __expf(-x * x);        // == exp(-z^2/2)
  float e = copysignf(erf_from_E(fabsf(x), E), x);
  return 0.5f * (1.0f + e) + z * E * (float)kInvSqrt2Pi;
}

// both at once, sharing E and the erf tail
__device__ __forceinline__ void gelu_and_grad(float z, float& g, float& dg) {
  float x = z * (float)kInvSqrt2;
  float E = __expf(-x * x);
  float e = copysignf(erf_from_E(fabsf(x), E), x);
  float phi = 0.5f * (1.0f + e);
  g = z * phi;
  dg = phi + z * E * (float)kInvSqrt2Pi;
}

__device__ __forceinline__ double gelu(double z) {
  return 0.5 * z * (1.0 + erf(z * kInvSqrt2));
}

__device__ __forceinline__ double gelu_grad(double z) {
  return 0.5 * (1.0 + erf(z * kInvSqrt2)) +
         z * exp(-0.5 * z * z) * kInvSqrt2Pi;
}

__device__ __forceinline__ void gelu_and_grad(double z, double& g, double& dg) {
  g = gelu(z);
  dg = gelu_grad(z);
}

}  // namespace dfno_gelu
