// RMSNorm kernels (bf16, fp32 accumulation) for CDNA4.
//
// Capability parity: the fused rmsnorm+residual epilogue the reference
// delegates to its vLLM containers (SURVEY.md §2.3 "RMSNorm ... fused ops").
//
// One 256-thread workgroup per token row; bf16x8 vector loads (16 B/lane);
// row held in registers between the sum-of-squares pass and the scale pass
// (template ITERS gives compile-time register indexing — guide §5.4 rule 20).

#include "common.h"

namespace fi {

// ITERS = ceil(hidden / (256 threads * 8 elems)); supports hidden <= ITERS*2048.
// FP8_OUT: `out` is u8 (OCP e4m3) and a per-row dynamic scale is written to
// `out_scales` — the fused epilogue for the fp8 serving mode (the normalized
// row only feeds a fp8 GEMM there, so the bf16 intermediate is never
// materialized).
template <int ITERS, bool FUSED_ADD, bool FP8_OUT>
__global__ void rms_norm_kernel(
    void* __restrict__ out_p,       // [T, H] bf16 (FP8_OUT: u8 e4m3)
    float* __restrict__ out_scales, // [T] (FP8_OUT only; else nullptr)
    const u16* __restrict__ in,     // [T, H]  (FUSED_ADD: the just-computed layer output x)
    u16* __restrict__ residual,     // [T, H]  in/out (FUSED_ADD only; else nullptr)
    const u16* __restrict__ weight, // [H]
    const float eps,
    const int hidden) {
  const int row = blockIdx.x;
  const int tid = threadIdx.x;
  const int64_t base = static_cast<int64_t>(row) * hidden;

  float vals[ITERS][8];
  float sumsq = 0.f;

#pragma unroll
  for (int it = 0; it < ITERS; ++it) {
    const int col = (it * blockDim.x + tid) * 8;
    if (col < hidden) {
      bf16x8 v = *reinterpret_cast<const bf16x8*>(in + base + col);
      if (FUSED_ADD) {
        bf16x8 r = *reinterpret_cast<const bf16x8*>(residual + base + col);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          vals[it][j] = bf16_to_f32(v.h[j]) + bf16_to_f32(r.h[j]);
        bf16x8 nr;
#pragma unroll
        for (int j = 0; j < 8; ++j) nr.h[j] = f32_to_bf16(vals[it][j]);
        *reinterpret_cast<bf16x8*>(residual + base + col) = nr;
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) vals[it][j] = bf16_to_f32(v.h[j]);
      }
#pragma unroll
      for (int j = 0; j < 8; ++j) sumsq += vals[it][j] * vals[it][j];
    }
  }

  __shared__ float scratch[4];
  sumsq = wave_reduce_sum(sumsq);
  sumsq = block_reduce_sum<4>(sumsq, scratch);
  const float inv_rms = rsqrtf(sumsq / hidden + eps);

  if (FP8_OUT) {
    // normalize into registers, then a second block reduce for the row
    // absmax -> per-row scale -> pack via v_cvt_pk_fp8_f32
    float amax = 0.f;
#pragma unroll
    for (int it = 0; it < ITERS; ++it) {
      const int col = (it * blockDim.x + tid) * 8;
      if (col < hidden) {
        bf16x8 w = *reinterpret_cast<const bf16x8*>(weight + col);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          vals[it][j] = vals[it][j] * inv_rms * bf16_to_f32(w.h[j]);
          amax = fmaxf(amax, fabsf(vals[it][j]));
        }
      }
    }
    __shared__ float scratch_max[4];
    amax = wave_reduce_max(amax);
    amax = block_reduce_max<4>(amax, scratch_max);
    const float scale = fmaxf(amax, 1e-8f) / 448.0f;
    const float inv_scale = 1.f / scale;
    if (tid == 0) out_scales[blockIdx.x] = scale;
    unsigned char* out8 = static_cast<unsigned char*>(out_p);
#pragma unroll
    for (int it = 0; it < ITERS; ++it) {
      const int col = (it * blockDim.x + tid) * 8;
      if (col < hidden) {
        float sv[8];
#pragma unroll
        for (int j = 0; j < 8; ++j) sv[j] = vals[it][j] * inv_scale;
        u32 packed[2];
        pack_fp8x8(sv, packed);
        *reinterpret_cast<u32*>(out8 + base + col) = packed[0];
        *reinterpret_cast<u32*>(out8 + base + col + 4) = packed[1];
      }
    }
  } else {
    u16* out = static_cast<u16*>(out_p);
#pragma unroll
    for (int it = 0; it < ITERS; ++it) {
      const int col = (it * blockDim.x + tid) * 8;
      if (col < hidden) {
        bf16x8 w = *reinterpret_cast<const bf16x8*>(weight + col);
        bf16x8 o;
#pragma unroll
        for (int j = 0; j < 8; ++j)
          o.h[j] = f32_to_bf16(vals[it][j] * inv_rms * bf16_to_f32(w.h[j]));
        *reinterpret_cast<bf16x8*>(out + base + col) = o;
      }
    }
  }
}

template <bool FUSED_ADD, bool FP8_OUT>
void launch_rms_norm(void* out, float* out_scales, const u16* in,
                     u16* residual, const u16* weight, float eps, int tokens,
                     int hidden, hipStream_t stream) {
  dim3 grid(tokens), block(256);
  const int iters = ceil_div(hidden, 256 * 8);
#define FI_CASE(N)                                                           \
  case N:                                                                    \
    hipLaunchKernelGGL((rms_norm_kernel<N, FUSED_ADD, FP8_OUT>), grid,       \
                       block, 0, stream, out, out_scales, in, residual,      \
                       weight, eps, hidden);                                 \
    break;
  switch (iters) {
    FI_CASE(1)
    FI_CASE(2)
    FI_CASE(3)
    FI_CASE(4)
    FI_CASE(5)
    FI_CASE(6)
    FI_CASE(7)
    FI_CASE(8)
    default:
      // hidden > 16384 unsupported by this kernel family
      abort();
  }
#undef FI_CASE
}

template void launch_rms_norm<true, false>(void*, float*, const u16*, u16*,
                                           const u16*, float, int, int,
                                           hipStream_t);
template void launch_rms_norm<false, false>(void*, float*, const u16*, u16*,
                                            const u16*, float, int, int,
                                            hipStream_t);
template void launch_rms_norm<true, true>(void*, float*, const u16*, u16*,
                                          const u16*, float, int, int,
                                          hipStream_t);
template void launch_rms_norm<false, true>(void*, float*, const u16*, u16*,
                                           const u16*, float, int, int,
                                           hipStream_t);

}  // namespace fi
