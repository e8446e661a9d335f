// FP8 (OCP e4m3) fused activation-quantization kernels for CDNA4.
//
// The fp8 serving mode needs per-token-quantized activations for
// torch._scaled_mm; a standalone eager quant pass measured MORE expensive
// than the fp8 GEMM savings, so quantization is fused into the producers:
//  * silu_and_mul_fp8: SwiGLU epilogue emitting fp8 + per-row scale
//    (feeds down_proj).
//  * quant_fp8_rows: bf16 -> fp8 row quant (feeds o_proj from the
//    attention output).
// The RMSNorm fp8 variants live in norm.hip (feed qkv/gate_up).
// Native v_cvt_pk_fp8_f32 packs 2 f32 -> 2 fp8 bytes per instruction.

#include "common.h"

namespace fi {

constexpr float kFp8Max = 448.0f;

// one 256-thread workgroup per row; row held in registers between the
// absmax pass and the convert pass (ITERS = ceil(cols / 2048))
template <int ITERS, bool SILU_MUL>
__global__ void row_quant_fp8_kernel(
    unsigned char* __restrict__ out,   // [T, cols] fp8
    float* __restrict__ out_scales,    // [T]
    const u16* __restrict__ in,        // [T, cols] bf16 (SILU_MUL: [T, 2*cols])
    const int cols) {
  const int row = blockIdx.x;
  const int tid = threadIdx.x;
  const int64_t in_row = static_cast<int64_t>(row) * (SILU_MUL ? 2 * cols : cols);

  float vals[ITERS][8];
  float amax = 0.f;
#pragma unroll
  for (int it = 0; it < ITERS; ++it) {
    const int col = (it * blockDim.x + tid) * 8;
    if (col < cols) {
      if (SILU_MUL) {
        bf16x8 g = *reinterpret_cast<const bf16x8*>(in + in_row + col);
        bf16x8 u = *reinterpret_cast<const bf16x8*>(in + in_row + cols + col);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const float gf = bf16_to_f32(g.h[j]);
          vals[it][j] = gf / (1.f + __expf(-gf)) * bf16_to_f32(u.h[j]);
        }
      } else {
        bf16x8 v = *reinterpret_cast<const bf16x8*>(in + in_row + col);
#pragma unroll
        for (int j = 0; j < 8; ++j) vals[it][j] = bf16_to_f32(v.h[j]);
      }
#pragma unroll
      for (int j = 0; j < 8; ++j) amax = fmaxf(amax, fabsf(vals[it][j]));
    }
  }

  __shared__ float scratch[4];
  amax = wave_reduce_max(amax);
  amax = block_reduce_max<4>(amax, scratch);
  const float scale = fmaxf(amax, 1e-8f) / kFp8Max;
  const float inv_scale = 1.f / scale;
  if (tid == 0) out_scales[row] = scale;

#pragma unroll
  for (int it = 0; it < ITERS; ++it) {
    const int col = (it * blockDim.x + tid) * 8;
    if (col < cols) {
      float sv[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) sv[j] = vals[it][j] * inv_scale;
      u32 packed[2];
      pack_fp8x8(sv, packed);
      *reinterpret_cast<u32*>(out + static_cast<int64_t>(row) * cols + col) =
          packed[0];
      *reinterpret_cast<u32*>(out + static_cast<int64_t>(row) * cols + col + 4) =
          packed[1];
    }
  }
}

template <bool SILU_MUL>
void launch_row_quant_fp8(unsigned char* out, float* scales, const u16* in,
                          int rows, int cols, hipStream_t stream) {
  dim3 grid(rows), block(256);
  const int iters = ceil_div(cols, 256 * 8);
#define FI_QCASE(N)                                                        \
  case N:                                                                  \
    hipLaunchKernelGGL((row_quant_fp8_kernel<N, SILU_MUL>), grid, block, 0, \
                       stream, out, scales, in, cols);                     \
    break;
  switch (iters) {
    FI_QCASE(1)
    FI_QCASE(2)
    FI_QCASE(3)
    FI_QCASE(4)
    FI_QCASE(5)
    FI_QCASE(6)
    FI_QCASE(7)
    FI_QCASE(8)
    default:
      abort();
  }
#undef FI_QCASE
}

template void launch_row_quant_fp8<true>(unsigned char*, float*, const u16*,
                                         int, int, hipStream_t);
template void launch_row_quant_fp8<false>(unsigned char*, float*, const u16*,
                                          int, int, hipStream_t);

}  // namespace fi
