// Common device helpers for FusionInfer-AMD CDNA4 (gfx950) kernels.
//
// Design notes (see /opt/skills/guides/cdna_hip_programming.md):
//  * wave64 everywhere; block sizes are multiples of 64.
//  * bf16 loads are ALWAYS vectorized (short4/short8 reinterpret, G13).
//  * bf16<->f32 via bit ops (RNE) — matches hardware v_cvt rounding.
#pragma once

#include <hip/hip_runtime.h>
#include <stdint.h>
#include <algorithm>

#define FI_DEV __device__ __forceinline__

namespace fi {

constexpr int kWaveSize = 64;

using u16 = unsigned short;
using u32 = unsigned int;

// 16-byte vector of 8 bf16 values (raw bits).
struct alignas(16) bf16x8 {
  u16 h[8];
};
// 8-byte vector of 4 bf16 values.
struct alignas(8) bf16x4 {
  u16 h[4];
};

FI_DEV float bf16_to_f32(u16 h) {
  union {
    u32 u;
    float f;
  } x;
  x.u = static_cast<u32>(h) << 16;
  return x.f;
}

FI_DEV u16 f32_to_bf16(float f) {
  // native cast -> ONE v_cvt_pk_bf16_f32 (RNE, NaN-safe); the manual
  // bit-twiddled RNE this replaces cost ~6 VALU ops per element
  union {
    u16 u;
    __bf16 h;
  } c;
  c.h = static_cast<__bf16>(f);
  return c.u;
}

// Wave-wide reductions (64 lanes).
FI_DEV float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}

FI_DEV float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, 64));
  return v;
}

// Block-wide sum over NWAVES waves (each wave passes its wave-reduced value
// with lane 0 valid). `scratch` must hold NWAVES floats.
template <int NWAVES>
FI_DEV float block_reduce_sum(float wave_val, float* scratch) {
  const int wave = threadIdx.x / kWaveSize;
  const int lane = threadIdx.x % kWaveSize;
  if (lane == 0) scratch[wave] = wave_val;
  __syncthreads();
  float total = 0.f;
#pragma unroll
  for (int i = 0; i < NWAVES; ++i) total += scratch[i];
  return total;
}

__host__ __device__ __forceinline__ constexpr int ceil_div(int a, int b) {
  return (a + b - 1) / b;
}

// pack 8 f32 into 8 OCP-e4m3 bytes (two u32 words) via v_cvt_pk_fp8_f32
FI_DEV void pack_fp8x8(const float* v, unsigned int out[2]) {
#if defined(__HIP_DEVICE_COMPILE__)
  unsigned int w0 = 0, w1 = 0;
  w0 = __builtin_amdgcn_cvt_pk_fp8_f32(v[0], v[1], w0, false);
  w0 = __builtin_amdgcn_cvt_pk_fp8_f32(v[2], v[3], w0, true);
  w1 = __builtin_amdgcn_cvt_pk_fp8_f32(v[4], v[5], w1, false);
  w1 = __builtin_amdgcn_cvt_pk_fp8_f32(v[6], v[7], w1, true);
  out[0] = w0;
  out[1] = w1;
#else
  out[0] = out[1] = 0;  // host pass never executes device code
#endif
}

// unpack 4 OCP-e4m3 bytes (one u32) -> 4 f32 via v_cvt_pk_f32_fp8
FI_DEV void unpack_fp8x4(u32 w, float out[4]) {
#if defined(__HIP_DEVICE_COMPILE__)
  typedef __attribute__((ext_vector_type(2))) float f32x2;
  const f32x2 lo = __builtin_amdgcn_cvt_pk_f32_fp8(w, false);
  const f32x2 hi = __builtin_amdgcn_cvt_pk_f32_fp8(w, true);
  out[0] = lo[0];
  out[1] = lo[1];
  out[2] = hi[0];
  out[3] = hi[1];
#else
  out[0] = out[1] = out[2] = out[3] = 0.f;
#endif
}

template <int NWAVES>
FI_DEV float block_reduce_max(float wave_val, float* scratch) {
  const int wave = threadIdx.x / kWaveSize;
  const int lane = threadIdx.x % kWaveSize;
  if (lane == 0) scratch[wave] = wave_val;
  __syncthreads();
  float m = scratch[0];
#pragma unroll
  for (int i = 1; i < NWAVES; ++i) m = fmaxf(m, scratch[i]);
  return m;
}

}  // namespace fi
