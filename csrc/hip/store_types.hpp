// Factor STORAGE types for the MTTKRP gather path: compute stays V
// (f64/f32), storage S may be narrower (f32/bf16) to cut gathered cache
// lines on HBM-bound shapes (ROADMAP 2b). Shared by mttkrp_flat.hip and
// mttkrp_lds.hip.
#pragma once
#include <hip/hip_runtime.h>
#include <cstdint>

namespace splatt_store {

struct bf16 { uint16_t v; };

template <typename S> struct VecElemT { using type = S; };
template <> struct VecElemT<bf16> { using type = uint16_t; };

__device__ __forceinline__ double to_compute(double x, double) { return x; }
__device__ __forceinline__ float to_compute(float x, float) { return x; }
__device__ __forceinline__ double to_compute(float x, double) {
  return (double)x;
}
__device__ __forceinline__ double to_compute(bf16 x, double) {
  union { uint32_t u; float f; } c;
  c.u = (uint32_t)x.v << 16;
  return (double)c.f;
}
__device__ __forceinline__ float to_compute(bf16 x, float) {
  union { uint32_t u; float f; } c;
  c.u = (uint32_t)x.v << 16;
  return c.f;
}

}  // namespace splatt_store
