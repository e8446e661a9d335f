// Fused SiLU-and-multiply (SwiGLU epilogue) for CDNA4.
// in: [T, 2*I] (gate | up) bf16 -> out: [T, I] bf16, fp32 math.
// Grid-stride over vectors of 8; memory-bound — bf16x8 loads (G13).

#include "common.h"

namespace fi {

__global__ void silu_and_mul_kernel(u16* __restrict__ out,
                                    const u16* __restrict__ in,
                                    const int64_t num_tokens,
                                    const int inter) {  // I (elements)
  const int vec_per_row = inter / 8;
  const int64_t total = num_tokens * vec_per_row;
  for (int64_t idx = blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += static_cast<int64_t>(gridDim.x) * blockDim.x) {
    const int64_t t = idx / vec_per_row;
    const int c = (idx % vec_per_row) * 8;
    const int64_t row_base = t * (2 * inter);
    bf16x8 g = *reinterpret_cast<const bf16x8*>(in + row_base + c);
    bf16x8 u = *reinterpret_cast<const bf16x8*>(in + row_base + inter + c);
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float gf = bf16_to_f32(g.h[j]);
      const float uf = bf16_to_f32(u.h[j]);
      const float s = gf / (1.f + __expf(-gf));
      o.h[j] = f32_to_bf16(s * uf);
    }
    *reinterpret_cast<bf16x8*>(out + t * inter + c) = o;
  }
}

void launch_silu_and_mul(u16* out, const u16* in, int64_t tokens, int inter,
                         hipStream_t stream) {
  const int64_t total = tokens * (inter / 8);
  const int block = 256;
  const int grid =
      static_cast<int>(std::min<int64_t>((total + block - 1) / block, (int64_t)2048));
  hipLaunchKernelGGL(silu_and_mul_kernel, dim3(grid), dim3(block), 0, stream,
                     out, in, tokens, inter);
}

}  // namespace fi
