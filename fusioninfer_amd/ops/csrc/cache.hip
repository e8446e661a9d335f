// Paged-KV cache kernels for CDNA4:
//  * reshape_and_cache — scatter freshly-computed K/V rows into the paged
//    cache ([num_blocks, Hk, block_size, D] bf16) by slot_mapping.
//  * gather/scatter_kv_blocks — pack cache blocks into ONE contiguous
//    staging buffer (and back) so the PD prefiller→decoder handoff is a
//    single large RCCL send over an xGMI link (SURVEY.md §2.3 "KV-block
//    pack/unpack for PD transfer"; §5.8 — pack into large contiguous
//    sends to hit link peak).

#include "common.h"

namespace fi {

// FP8: the cache stores OCP e4m3 at a per-layer static scale (default
// 1.0: K rows are qk-normed and O(1); saturation at |448| is the only
// loss) — halves decode KV bytes and doubles cache capacity. The write
// multiplies by the INVERSE scale; the read side needs no kernel work
// (K's scale folds into the softmax scalar, V's into a post-multiply on
// the attention output). Conversion uses v_cvt_pk_fp8_f32.
template <bool FP8>
__global__ void reshape_and_cache_kernel(
    const u16* __restrict__ k,   // [T] rows of Hk*D, stride k_stride (bf16)
    const u16* __restrict__ v,
    void* __restrict__ k_cache,  // [B, Hk, bs, D] bf16 (FP8: e4m3 bytes)
    void* __restrict__ v_cache,
    const int* __restrict__ slot_mapping,  // [T]
    const int64_t k_stride, const int64_t v_stride,
    const int num_tokens, const int kv_heads, const int block_size,
    const int head_dim, const float k_inv_scale, const float v_inv_scale) {
  const int vec_per_tok = kv_heads * head_dim / 8;
  const int64_t total = static_cast<int64_t>(num_tokens) * vec_per_tok;
  for (int64_t idx = blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += static_cast<int64_t>(gridDim.x) * blockDim.x) {
    const int t = static_cast<int>(idx / vec_per_tok);
    const int c = static_cast<int>(idx % vec_per_tok) * 8;  // elem in [0,Hk*D)
    const int slot = slot_mapping[t];
    if (slot < 0) continue;  // padding token
    const int blk = slot / block_size;
    const int off = slot % block_size;
    const int h = c / head_dim;
    const int d = c % head_dim;
    const int64_t dst =
        ((static_cast<int64_t>(blk) * kv_heads + h) * block_size + off) *
            head_dim + d;
    const bf16x8 kv = *reinterpret_cast<const bf16x8*>(k + t * k_stride + c);
    const bf16x8 vv = *reinterpret_cast<const bf16x8*>(v + t * v_stride + c);
    if (FP8) {
      float kf[8], vf[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        // saturate at e4m3 max: v_cvt_pk_fp8_f32 NaNs out-of-range
        // values (no hardware clamp), and a NaN in the cache poisons
        // every later attention read of the block
        kf[j] = fminf(fmaxf(bf16_to_f32(kv.h[j]) * k_inv_scale, -448.f),
                      448.f);
        vf[j] = fminf(fmaxf(bf16_to_f32(vv.h[j]) * v_inv_scale, -448.f),
                      448.f);
      }
      u32 kp[2], vp[2];
      pack_fp8x8(kf, kp);
      pack_fp8x8(vf, vp);
      *reinterpret_cast<uint2*>(static_cast<unsigned char*>(k_cache) + dst) =
          make_uint2(kp[0], kp[1]);
      *reinterpret_cast<uint2*>(static_cast<unsigned char*>(v_cache) + dst) =
          make_uint2(vp[0], vp[1]);
    } else {
      *reinterpret_cast<bf16x8*>(static_cast<u16*>(k_cache) + dst) = kv;
      *reinterpret_cast<bf16x8*>(static_cast<u16*>(v_cache) + dst) = vv;
    }
  }
}

void launch_reshape_and_cache(const u16* k, const u16* v, void* k_cache,
                              void* v_cache, const int* slot_mapping,
                              int64_t k_stride, int64_t v_stride, int tokens,
                              int kv_heads, int block_size, int head_dim,
                              bool fp8, float k_inv_scale, float v_inv_scale,
                              hipStream_t stream) {
  const int64_t total = static_cast<int64_t>(tokens) * kv_heads * head_dim / 8;
  const int block = 256;
  const int grid = static_cast<int>(std::min<int64_t>((total + block - 1) / block, (int64_t)2048));
  if (fp8) {
    hipLaunchKernelGGL((reshape_and_cache_kernel<true>), dim3(grid),
                       dim3(block), 0, stream, k, v, k_cache, v_cache,
                       slot_mapping, k_stride, v_stride, tokens, kv_heads,
                       block_size, head_dim, k_inv_scale, v_inv_scale);
  } else {
    hipLaunchKernelGGL((reshape_and_cache_kernel<false>), dim3(grid),
                       dim3(block), 0, stream, k, v, k_cache, v_cache,
                       slot_mapping, k_stride, v_stride, tokens, kv_heads,
                       block_size, head_dim, 1.0f, 1.0f);
  }
}

// staging: [2, n, Hk, bs, D]; GATHER ? cache->staging : staging->cache.
template <bool GATHER>
__global__ void kv_block_copy_kernel(u16* __restrict__ staging,
                                     u16* __restrict__ k_cache,
                                     u16* __restrict__ v_cache,
                                     const int* __restrict__ block_ids,
                                     const int num_blocks,
                                     const int64_t block_elems) {  // Hk*bs*D
  const int64_t vec_per_block = block_elems / 8;
  const int64_t total = num_blocks * vec_per_block;
  for (int64_t idx = blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += static_cast<int64_t>(gridDim.x) * blockDim.x) {
    const int i = static_cast<int>(idx / vec_per_block);
    const int64_t c = (idx % vec_per_block) * 8;
    const int64_t cache_off = static_cast<int64_t>(block_ids[i]) * block_elems + c;
    const int64_t stage_off = static_cast<int64_t>(i) * block_elems + c;
    if (GATHER) {
      *reinterpret_cast<bf16x8*>(staging + stage_off) =
          *reinterpret_cast<const bf16x8*>(k_cache + cache_off);
      *reinterpret_cast<bf16x8*>(staging + num_blocks * block_elems + stage_off) =
          *reinterpret_cast<const bf16x8*>(v_cache + cache_off);
    } else {
      *reinterpret_cast<bf16x8*>(k_cache + cache_off) =
          *reinterpret_cast<const bf16x8*>(staging + stage_off);
      *reinterpret_cast<bf16x8*>(v_cache + cache_off) =
          *reinterpret_cast<const bf16x8*>(
              staging + num_blocks * block_elems + stage_off);
    }
  }
}

template <bool GATHER>
void launch_kv_block_copy(u16* staging, u16* k_cache, u16* v_cache,
                          const int* block_ids, int num_blocks,
                          int64_t block_elems, hipStream_t stream) {
  const int64_t total = num_blocks * (block_elems / 8);
  const int block = 256;
  const int grid = static_cast<int>(std::min<int64_t>((total + block - 1) / block, (int64_t)4096));
  hipLaunchKernelGGL((kv_block_copy_kernel<GATHER>), dim3(grid), dim3(block),
                     0, stream, staging, k_cache, v_cache, block_ids,
                     num_blocks, block_elems);
}

template void launch_kv_block_copy<true>(u16*, u16*, u16*, const int*, int,
                                         int64_t, hipStream_t);
template void launch_kv_block_copy<false>(u16*, u16*, u16*, const int*, int,
                                          int64_t, hipStream_t);

}  // namespace fi
