// Fused per-head RMSNorm (Qwen3 qk-norm) + NeoX rotary embedding, in-place.
//
// One wave per (token, head): D/2 lanes hold 2 bf16 each; the rotate-half
// partner lives exactly shfl_xor(D/4) away, so RoPE needs one cross-lane
// exchange and zero LDS. cos/sin are precomputed on host (f32 [P, D],
// cos | sin) — on-device trig would turn this memory-bound op VALU-bound
// (guide Appendix B).
//
// Capability parity: RoPE/qk-norm epilogue the reference delegates to vLLM
// (SURVEY.md §2.3).

#include "common.h"

namespace fi {

template <int D>
__global__ void rope_qk_norm_kernel(
    u16* __restrict__ q,             // [T, Hq*D] rows, stride q_stride
    u16* __restrict__ k,             // [T, Hk*D] rows, stride k_stride
    const int64_t q_stride, const int64_t k_stride,
    const u16* __restrict__ q_weight,  // [D] or nullptr
    const u16* __restrict__ k_weight,  // [D] or nullptr
    const float* __restrict__ cos_sin,  // [P, D]
    const int* __restrict__ positions,  // [T]
    const int num_q_heads, const float eps) {
  constexpr int kHalfLanes = D / 4;   // lanes holding x1 pairs
  constexpr int kActive = D / 2;
  const int lane = threadIdx.x;
  if (lane >= kActive) return;

  const int token = blockIdx.x;
  const int head = blockIdx.y;
  const bool is_q = head < num_q_heads;
  u16* base = is_q ? q + token * q_stride + static_cast<int64_t>(head) * D
                   : k + token * k_stride +
                         static_cast<int64_t>(head - num_q_heads) * D;
  const u16* w = is_q ? q_weight : k_weight;

  // element offset of this lane's pair within the head
  const int e = 2 * lane;
  u32 bits = *reinterpret_cast<const u32*>(base + e);
  float a = bf16_to_f32(static_cast<u16>(bits & 0xffff));
  float b = bf16_to_f32(static_cast<u16>(bits >> 16));

  if (w != nullptr) {
    float sumsq = a * a + b * b;
#pragma unroll
    for (int off = kActive / 2; off > 0; off >>= 1)
      sumsq += __shfl_xor(sumsq, off, 64);
    const float inv_rms = rsqrtf(sumsq / D + eps);
    const u32 wbits = *reinterpret_cast<const u32*>(w + e);
    a *= inv_rms * bf16_to_f32(static_cast<u16>(wbits & 0xffff));
    b *= inv_rms * bf16_to_f32(static_cast<u16>(wbits >> 16));
  }

  // rotate-half partner exchange
  const float pa = __shfl_xor(a, kHalfLanes, 64);
  const float pb = __shfl_xor(b, kHalfLanes, 64);

  const bool is_x1 = lane < kHalfLanes;
  const int pair = is_x1 ? e : e - D / 2;  // rotary pair index in [0, D/2)
  const int pos = positions[token];
  const float2 c = *reinterpret_cast<const float2*>(
      cos_sin + static_cast<int64_t>(pos) * D + pair);
  const float2 s = *reinterpret_cast<const float2*>(
      cos_sin + static_cast<int64_t>(pos) * D + D / 2 + pair);

  float oa, ob;
  if (is_x1) {  // x1' = x1*cos - x2*sin
    oa = a * c.x - pa * s.x;
    ob = b * c.y - pb * s.y;
  } else {  // x2' = x2*cos + x1*sin
    oa = a * c.x + pa * s.x;
    ob = b * c.y + pb * s.y;
  }
  u32 obits = static_cast<u32>(f32_to_bf16(oa)) |
              (static_cast<u32>(f32_to_bf16(ob)) << 16);
  *reinterpret_cast<u32*>(base + e) = obits;
}

void launch_rope_qk_norm(u16* q, u16* k, int64_t q_stride, int64_t k_stride,
                         const u16* q_weight, const u16* k_weight,
                         const float* cos_sin, const int* positions,
                         int tokens, int num_q_heads, int num_kv_heads,
                         int head_dim, float eps, hipStream_t stream) {
  dim3 grid(tokens, num_q_heads + num_kv_heads), block(64);
  if (head_dim == 128) {
    hipLaunchKernelGGL((rope_qk_norm_kernel<128>), grid, block, 0, stream, q,
                       k, q_stride, k_stride, q_weight, k_weight, cos_sin,
                       positions, num_q_heads, eps);
  } else if (head_dim == 64) {
    hipLaunchKernelGGL((rope_qk_norm_kernel<64>), grid, block, 0, stream, q, k,
                       q_stride, k_stride, q_weight, k_weight, cos_sin,
                       positions, num_q_heads, eps);
  } else {
    abort();  // head_dim ∈ {64, 128}
  }
}

}  // namespace fi
