// PyTorch bindings for the FusionInfer-AMD CDNA4 kernel library.

#include <torch/extension.h>

#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

#include "common.h"

namespace fi {

template <bool FUSED_ADD, bool FP8_OUT>
void launch_rms_norm(void*, float*, const u16*, u16*, const u16*, float, int,
                     int, hipStream_t);
void launch_silu_and_mul(u16*, const u16*, int64_t, int, hipStream_t);
template <bool SILU_MUL>
void launch_row_quant_fp8(unsigned char*, float*, const u16*, int, int,
                          hipStream_t);
void launch_rope_qk_norm(u16*, u16*, int64_t, int64_t, const u16*, const u16*,
                         const float*, const int*, int, int, int, int, float,
                         hipStream_t);
void launch_reshape_and_cache(const u16*, const u16*, void*, void*,
                              const int*, int64_t, int64_t, int, int, int,
                              int, bool, float, float, hipStream_t);
template <bool GATHER>
void launch_kv_block_copy(u16*, u16*, u16*, const int*, int, int64_t,
                          hipStream_t);
void launch_paged_attn_decode(u16*, float*, float*, const u16*, const void*,
                              const void*, const int*, const int*, int,
                              int64_t, int, int, int, int, int, float, bool,
                              hipStream_t);
void launch_prefill_attn(u16*, const u16*, const void*, const void*,
                         const int*, const int*, const int*, const int*,
                         const int*, int, int, int64_t, int64_t, int64_t, int,
                         int, int, float, bool, int, hipStream_t);
void launch_moe_gemm(u16*, const u16*, const u16*, const int*, const int*,
                     const int*, int, int, int, int, bool, hipStream_t);
void launch_moe_gemm_fp8(u16*, const unsigned char*, const float*,
                         const unsigned char*, const float*, const int*,
                         const int*, const int*, int, int, int, int, bool,
                         hipStream_t);
void launch_moe_align(const int*, int*, int*, int*, int*, int, int, int,
                      int, int, int, hipStream_t);
void launch_moe_router_topk(const float*, float*, int*, int, int, int,
                            bool, hipStream_t);
void launch_moe_combine(u16*, const u16*, const int*, const float*, int, int,
                        int, hipStream_t);

}  // namespace fi

namespace {

using fi::u16;

#define CHECK_BF16_CUDA(t)                                     \
  TORCH_CHECK((t).is_cuda(), #t " must be on GPU");            \
  TORCH_CHECK((t).scalar_type() == at::kBFloat16, #t " must be bf16")

u16* bf16_ptr(at::Tensor& t) { return reinterpret_cast<u16*>(t.data_ptr()); }
const u16* bf16_cptr(const at::Tensor& t) {
  return reinterpret_cast<const u16*>(t.data_ptr());
}

hipStream_t current_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

// KV caches are bf16 or fp8 (OCP e4m3, scale 1.0)
bool cache_is_fp8(const at::Tensor& t) {
  TORCH_CHECK(
      t.scalar_type() == at::kBFloat16 || t.scalar_type() == at::kFloat8_e4m3fn,
      "kv cache must be bf16 or float8_e4m3fn");
  return t.scalar_type() == at::kFloat8_e4m3fn;
}

void rms_norm(at::Tensor out, at::Tensor input, at::Tensor weight, double eps) {
  CHECK_BF16_CUDA(out);
  CHECK_BF16_CUDA(input);
  TORCH_CHECK(input.is_contiguous() && out.is_contiguous());
  const int hidden = input.size(-1);
  const int tokens = input.numel() / hidden;
  fi::launch_rms_norm<false, false>(bf16_ptr(out), nullptr, bf16_cptr(input),
                                    nullptr, bf16_cptr(weight),
                                    static_cast<float>(eps), tokens, hidden,
                                    current_stream());
}

void fused_add_rms_norm(at::Tensor input, at::Tensor residual,
                        at::Tensor weight, double eps) {
  CHECK_BF16_CUDA(input);
  CHECK_BF16_CUDA(residual);
  TORCH_CHECK(input.is_contiguous() && residual.is_contiguous());
  const int hidden = input.size(-1);
  const int tokens = input.numel() / hidden;
  // in-place: residual += input; input = rmsnorm(residual)
  fi::launch_rms_norm<true, false>(bf16_ptr(input), nullptr, bf16_cptr(input),
                                   bf16_ptr(residual), bf16_cptr(weight),
                                   static_cast<float>(eps), tokens, hidden,
                                   current_stream());
}

#define CHECK_FP8_OUT(o, s)                                                  \
  TORCH_CHECK((o).scalar_type() == at::kFloat8_e4m3fn && (o).is_cuda() &&    \
              (o).is_contiguous());                                          \
  TORCH_CHECK((s).scalar_type() == at::kFloat && (s).is_cuda())

void rms_norm_fp8(at::Tensor out, at::Tensor out_scales, at::Tensor input,
                  at::Tensor weight, double eps) {
  CHECK_BF16_CUDA(input);
  CHECK_FP8_OUT(out, out_scales);
  TORCH_CHECK(input.is_contiguous());
  const int hidden = input.size(-1);
  const int tokens = input.numel() / hidden;
  fi::launch_rms_norm<false, true>(
      out.data_ptr(), out_scales.data_ptr<float>(), bf16_cptr(input), nullptr,
      bf16_cptr(weight), static_cast<float>(eps), tokens, hidden,
      current_stream());
}

void fused_add_rms_norm_fp8(at::Tensor out, at::Tensor out_scales,
                            at::Tensor input, at::Tensor residual,
                            at::Tensor weight, double eps) {
  CHECK_BF16_CUDA(input);
  CHECK_BF16_CUDA(residual);
  CHECK_FP8_OUT(out, out_scales);
  TORCH_CHECK(input.is_contiguous() && residual.is_contiguous());
  const int hidden = input.size(-1);
  const int tokens = input.numel() / hidden;
  // residual += input (in place); out = fp8(rmsnorm(residual))
  fi::launch_rms_norm<true, true>(
      out.data_ptr(), out_scales.data_ptr<float>(), bf16_cptr(input),
      bf16_ptr(residual), bf16_cptr(weight), static_cast<float>(eps), tokens,
      hidden, current_stream());
}

void silu_and_mul_fp8(at::Tensor out, at::Tensor out_scales, at::Tensor input) {
  CHECK_BF16_CUDA(input);
  CHECK_FP8_OUT(out, out_scales);
  TORCH_CHECK(input.is_contiguous());
  const int inter = out.size(-1);
  TORCH_CHECK(input.size(-1) == 2 * inter);
  TORCH_CHECK(inter % 8 == 0);
  fi::launch_row_quant_fp8<true>(
      static_cast<unsigned char*>(out.data_ptr()),
      out_scales.data_ptr<float>(), bf16_cptr(input),
      input.numel() / (2 * inter), inter, current_stream());
}

void quant_fp8_rows(at::Tensor out, at::Tensor out_scales, at::Tensor input) {
  CHECK_BF16_CUDA(input);
  CHECK_FP8_OUT(out, out_scales);
  TORCH_CHECK(input.is_contiguous());
  const int cols = input.size(-1);
  TORCH_CHECK(cols % 8 == 0);
  fi::launch_row_quant_fp8<false>(
      static_cast<unsigned char*>(out.data_ptr()),
      out_scales.data_ptr<float>(), bf16_cptr(input), input.numel() / cols,
      cols, current_stream());
}

void silu_and_mul(at::Tensor out, at::Tensor input) {
  CHECK_BF16_CUDA(out);
  CHECK_BF16_CUDA(input);
  TORCH_CHECK(input.is_contiguous() && out.is_contiguous());
  const int inter = out.size(-1);
  TORCH_CHECK(input.size(-1) == 2 * inter);
  TORCH_CHECK(inter % 8 == 0);
  fi::launch_silu_and_mul(bf16_ptr(out), bf16_cptr(input),
                          input.numel() / (2 * inter), inter,
                          current_stream());
}

void rope_qk_norm(at::Tensor q, at::Tensor k,
                  c10::optional<at::Tensor> q_weight,
                  c10::optional<at::Tensor> k_weight, at::Tensor cos_sin,
                  at::Tensor positions, int64_t num_q_heads,
                  int64_t num_kv_heads, int64_t head_dim, double eps) {
  CHECK_BF16_CUDA(q);
  CHECK_BF16_CUDA(k);
  TORCH_CHECK(cos_sin.scalar_type() == at::kFloat && cos_sin.is_cuda());
  TORCH_CHECK(positions.scalar_type() == at::kInt && positions.is_cuda());
  // q/k may be row-strided slices of the fused qkv projection output
  TORCH_CHECK(q.dim() == 2 && k.dim() == 2);
  TORCH_CHECK(q.stride(1) == 1 && k.stride(1) == 1);
  const int tokens = q.size(0);
  fi::launch_rope_qk_norm(
      bf16_ptr(q), bf16_ptr(k), q.stride(0), k.stride(0),
      q_weight ? bf16_cptr(*q_weight) : nullptr,
      k_weight ? bf16_cptr(*k_weight) : nullptr, cos_sin.data_ptr<float>(),
      positions.data_ptr<int>(), tokens, num_q_heads, num_kv_heads, head_dim,
      static_cast<float>(eps), current_stream());
}

void reshape_and_cache(at::Tensor k, at::Tensor v, at::Tensor k_cache,
                       at::Tensor v_cache, at::Tensor slot_mapping,
                       double k_inv_scale, double v_inv_scale) {
  CHECK_BF16_CUDA(k);
  CHECK_BF16_CUDA(v);
  const bool fp8 = cache_is_fp8(k_cache);
  TORCH_CHECK(slot_mapping.scalar_type() == at::kInt);
  TORCH_CHECK(k.dim() == 2 && v.dim() == 2);  // [T, Hk*D] (maybe strided rows)
  TORCH_CHECK(k.stride(1) == 1 && v.stride(1) == 1);
  const int kv_heads = k_cache.size(1);
  const int block_size = k_cache.size(2);
  const int head_dim = k_cache.size(3);
  fi::launch_reshape_and_cache(
      bf16_cptr(k), bf16_cptr(v), k_cache.data_ptr(), v_cache.data_ptr(),
      slot_mapping.data_ptr<int>(), k.stride(0), v.stride(0), k.size(0),
      kv_heads, block_size, head_dim, fp8,
      static_cast<float>(k_inv_scale), static_cast<float>(v_inv_scale),
      current_stream());
}

void gather_kv_blocks(at::Tensor staging, at::Tensor k_cache,
                      at::Tensor v_cache, at::Tensor block_ids) {
  TORCH_CHECK(staging.is_cuda() &&
              staging.scalar_type() == k_cache.scalar_type());
  TORCH_CHECK(block_ids.scalar_type() == at::kInt);
  const int n = block_ids.size(0);
  // raw byte mover in u16 units (fp8 blocks are half the bytes)
  const int64_t block_elems = k_cache.size(1) * k_cache.size(2) *
      k_cache.size(3) * k_cache.element_size() / 2;
  fi::launch_kv_block_copy<true>(
      static_cast<fi::u16*>(staging.data_ptr()),
      static_cast<fi::u16*>(k_cache.data_ptr()),
      static_cast<fi::u16*>(v_cache.data_ptr()), block_ids.data_ptr<int>(), n,
      block_elems, current_stream());
}

void scatter_kv_blocks(at::Tensor staging, at::Tensor k_cache,
                       at::Tensor v_cache, at::Tensor block_ids) {
  TORCH_CHECK(staging.is_cuda() &&
              staging.scalar_type() == k_cache.scalar_type());
  TORCH_CHECK(block_ids.scalar_type() == at::kInt);
  const int n = block_ids.size(0);
  const int64_t block_elems = k_cache.size(1) * k_cache.size(2) *
      k_cache.size(3) * k_cache.element_size() / 2;
  fi::launch_kv_block_copy<false>(
      static_cast<fi::u16*>(staging.data_ptr()),
      static_cast<fi::u16*>(k_cache.data_ptr()),
      static_cast<fi::u16*>(v_cache.data_ptr()), block_ids.data_ptr<int>(), n,
      block_elems, current_stream());
}

void paged_attention_decode(at::Tensor out, at::Tensor q, at::Tensor k_cache,
                            at::Tensor v_cache, at::Tensor block_tables,
                            at::Tensor seq_lens,
                            c10::optional<at::Tensor> ml_ws,
                            c10::optional<at::Tensor> acc_ws,
                            int64_t num_parts, double scale) {
  CHECK_BF16_CUDA(out);
  CHECK_BF16_CUDA(q);
  const bool fp8 = cache_is_fp8(k_cache);
  TORCH_CHECK(block_tables.scalar_type() == at::kInt &&
              block_tables.is_contiguous());
  TORCH_CHECK(seq_lens.scalar_type() == at::kInt);
  TORCH_CHECK(q.dim() == 3);  // [S, Hq, D], rows may be strided
  const int num_seqs = q.size(0);
  const int num_heads = q.size(1);
  const int head_dim = q.size(2);
  const int num_kv_heads = k_cache.size(1);
  TORCH_CHECK(k_cache.size(2) == 16, "cache block_size must be 16");
  TORCH_CHECK(q.stride(2) == 1 && q.stride(1) == head_dim);
  float* ml = nullptr;
  float* acc = nullptr;
  if (num_parts > 1) {
    TORCH_CHECK(ml_ws && acc_ws, "workspace required when num_parts > 1");
    ml = ml_ws->data_ptr<float>();
    acc = acc_ws->data_ptr<float>();
  }
  fi::launch_paged_attn_decode(
      bf16_ptr(out), ml, acc, bf16_cptr(q), k_cache.data_ptr(),
      v_cache.data_ptr(), block_tables.data_ptr<int>(),
      seq_lens.data_ptr<int>(), num_seqs, q.stride(0), block_tables.size(1),
      num_kv_heads, head_dim, num_heads / num_kv_heads,
      static_cast<int>(num_parts), static_cast<float>(scale), fp8,
      current_stream());
}

void prefill_attention(at::Tensor out, at::Tensor q, at::Tensor k,
                       at::Tensor v, at::Tensor tile_seq, at::Tensor tile_row0,
                       at::Tensor cu_seqlens, double scale,
                       int64_t tile_rows) {
  TORCH_CHECK(tile_rows == 128 || tile_rows == 256,
              "tile_rows must be 128 or 256");
  CHECK_BF16_CUDA(out);
  CHECK_BF16_CUDA(q);
  TORCH_CHECK(tile_seq.scalar_type() == at::kInt &&
              tile_row0.scalar_type() == at::kInt &&
              cu_seqlens.scalar_type() == at::kInt);
  TORCH_CHECK(q.dim() == 3 && k.dim() == 3 && v.dim() == 3);
  const int num_q_heads = q.size(1);
  const int head_dim = q.size(2);
  const int num_kv_heads = k.size(1);
  TORCH_CHECK(q.stride(2) == 1 && q.stride(1) == head_dim);
  TORCH_CHECK(k.stride(2) == 1 && k.stride(1) == head_dim);
  TORCH_CHECK(v.stride(2) == 1 && v.stride(1) == head_dim);
  fi::launch_prefill_attn(
      bf16_ptr(out), bf16_cptr(q), bf16_cptr(k), bf16_cptr(v),
      tile_seq.data_ptr<int>(), tile_row0.data_ptr<int>(),
      cu_seqlens.data_ptr<int>(), nullptr, nullptr, 0, tile_seq.size(0),
      q.stride(0), k.stride(0), v.stride(0), num_q_heads, num_kv_heads,
      head_dim, static_cast<float>(scale), false,
      static_cast<int>(tile_rows / 32), current_stream());
}

void prefill_attention_paged(at::Tensor out, at::Tensor q, at::Tensor k_cache,
                             at::Tensor v_cache, at::Tensor tile_seq,
                             at::Tensor tile_row0, at::Tensor cu_seqlens,
                             at::Tensor block_tables, at::Tensor seq_lens_k,
                             double scale, int64_t tile_rows) {
  TORCH_CHECK(tile_rows == 128 || tile_rows == 256,
              "tile_rows must be 128 or 256");
  CHECK_BF16_CUDA(out);
  CHECK_BF16_CUDA(q);
  const bool fp8 = cache_is_fp8(k_cache);
  TORCH_CHECK(tile_seq.scalar_type() == at::kInt &&
              tile_row0.scalar_type() == at::kInt &&
              cu_seqlens.scalar_type() == at::kInt &&
              block_tables.scalar_type() == at::kInt &&
              seq_lens_k.scalar_type() == at::kInt);
  TORCH_CHECK(block_tables.is_contiguous());
  TORCH_CHECK(q.dim() == 3);
  const int num_q_heads = q.size(1);
  const int head_dim = q.size(2);
  const int num_kv_heads = k_cache.size(1);
  TORCH_CHECK(k_cache.size(2) == 16, "cache block_size must be 16");
  TORCH_CHECK(q.stride(2) == 1 && q.stride(1) == head_dim);
  fi::launch_prefill_attn(
      bf16_ptr(out), bf16_cptr(q), k_cache.data_ptr(), v_cache.data_ptr(),
      tile_seq.data_ptr<int>(), tile_row0.data_ptr<int>(),
      cu_seqlens.data_ptr<int>(), block_tables.data_ptr<int>(),
      seq_lens_k.data_ptr<int>(), block_tables.size(1), tile_seq.size(0),
      q.stride(0), 0, 0, num_q_heads, num_kv_heads, head_dim,
      static_cast<float>(scale), fp8,
      static_cast<int>(tile_rows / 32), current_stream());
}

void moe_gemm(at::Tensor out, at::Tensor a, at::Tensor b_packed,
              at::Tensor sorted_ids, at::Tensor expert_ids,
              at::Tensor n_valid, int64_t block_m, bool gate_up) {
  CHECK_BF16_CUDA(out);
  CHECK_BF16_CUDA(a);
  CHECK_BF16_CUDA(b_packed);
  TORCH_CHECK(out.is_contiguous() && a.is_contiguous() &&
              b_packed.is_contiguous());
  TORCH_CHECK(sorted_ids.scalar_type() == at::kInt &&
              expert_ids.scalar_type() == at::kInt &&
              n_valid.scalar_type() == at::kInt);
  TORCH_CHECK(block_m == 16 || block_m == 128, "block_m must be 16 or 128");
  const int K = a.size(1);
  const int N = out.size(1);
  const int PM = out.size(0);
  TORCH_CHECK(PM % block_m == 0, "padded rows must be block_m-aligned");
  TORCH_CHECK(K % 32 == 0, "K must be a multiple of 32");
  TORCH_CHECK(
      N % (block_m == 16 ? 64 : (gate_up ? 64 : 128)) == 0,
      "N tile misalignment");
  // packed B: [E, K/32, NB/16, 64, 8]
  TORCH_CHECK(b_packed.dim() == 5 && b_packed.size(1) == K / 32 &&
              b_packed.size(2) == (gate_up ? 2 * N : N) / 16 &&
              b_packed.size(3) == 64 && b_packed.size(4) == 8,
              "b_packed layout mismatch");
  TORCH_CHECK(expert_ids.size(0) >= PM / block_m);
  if (gate_up) TORCH_CHECK(sorted_ids.size(0) >= PM);
  fi::launch_moe_gemm(bf16_ptr(out), bf16_cptr(a),
                      bf16_cptr(b_packed), sorted_ids.data_ptr<int>(),
                      expert_ids.data_ptr<int>(), n_valid.data_ptr<int>(),
                      PM / block_m, K, N, static_cast<int>(block_m), gate_up,
                      current_stream());
}

void moe_gemm_fp8(at::Tensor out, at::Tensor a, at::Tensor a_scales,
                  at::Tensor b_packed, at::Tensor b_scales,
                  at::Tensor sorted_ids, at::Tensor expert_ids,
                  at::Tensor n_valid, int64_t block_m, bool gate_up) {
  CHECK_BF16_CUDA(out);
  TORCH_CHECK(a.scalar_type() == at::kFloat8_e4m3fn && a.is_cuda() &&
              a.is_contiguous());
  TORCH_CHECK(b_packed.scalar_type() == at::kFloat8_e4m3fn &&
              b_packed.is_contiguous());
  TORCH_CHECK(a_scales.scalar_type() == at::kFloat &&
              b_scales.scalar_type() == at::kFloat &&
              b_scales.is_contiguous());
  TORCH_CHECK(sorted_ids.scalar_type() == at::kInt &&
              expert_ids.scalar_type() == at::kInt &&
              n_valid.scalar_type() == at::kInt);
  TORCH_CHECK(block_m == 16 || block_m == 128);
  const int K = a.size(1);
  const int N = out.size(1);
  const int PM = out.size(0);
  TORCH_CHECK(PM % block_m == 0 && K % 32 == 0);
  TORCH_CHECK(N % (block_m == 16 ? 64 : (gate_up ? 64 : 128)) == 0);
  TORCH_CHECK(b_packed.dim() == 5 && b_packed.size(1) == K / 32 &&
              b_packed.size(2) == (gate_up ? 2 * N : N) / 16 &&
              b_packed.size(3) == 64 && b_packed.size(4) == 8);
  TORCH_CHECK(b_scales.numel() ==
              b_packed.size(0) * (gate_up ? 2 * N : N));
  fi::launch_moe_gemm_fp8(
      bf16_ptr(out), static_cast<const unsigned char*>(a.data_ptr()),
      a_scales.data_ptr<float>(),
      static_cast<const unsigned char*>(b_packed.data_ptr()),
      b_scales.data_ptr<float>(), sorted_ids.data_ptr<int>(),
      expert_ids.data_ptr<int>(), n_valid.data_ptr<int>(), PM / block_m, K,
      N, static_cast<int>(block_m), gate_up, current_stream());
}

void moe_combine(at::Tensor out, at::Tensor y, at::Tensor pos, at::Tensor w) {
  CHECK_BF16_CUDA(out);
  CHECK_BF16_CUDA(y);
  TORCH_CHECK(out.is_contiguous() && y.is_contiguous());
  TORCH_CHECK(pos.scalar_type() == at::kInt && pos.is_contiguous());
  TORCH_CHECK(w.scalar_type() == at::kFloat && w.is_contiguous());
  const int T = out.size(0);
  const int H = out.size(1);
  TORCH_CHECK(y.size(1) == H && H % 8 == 0);
  const int topk = pos.numel() / T;
  TORCH_CHECK(pos.numel() == T * topk && w.numel() == T * topk);
  fi::launch_moe_combine(bf16_ptr(out), bf16_cptr(y), pos.data_ptr<int>(),
                         w.data_ptr<float>(), T, topk, H, current_stream());
}

void moe_align(at::Tensor topi, at::Tensor sorted_ids, at::Tensor expert_ids,
               at::Tensor n_valid, at::Tensor pos, int64_t topk,
               int64_t e_start, int64_t e_end, int64_t block_m) {
  TORCH_CHECK(topi.is_cuda() && topi.scalar_type() == at::kInt &&
              topi.is_contiguous());
  TORCH_CHECK(sorted_ids.scalar_type() == at::kInt &&
              expert_ids.scalar_type() == at::kInt &&
              n_valid.scalar_type() == at::kInt &&
              pos.scalar_type() == at::kInt);
  const int n = topi.numel();
  const int PM = sorted_ids.numel();
  TORCH_CHECK(e_end - e_start <= 128, "moe_align: E_local > 128");
  TORCH_CHECK(PM % block_m == 0 && expert_ids.numel() == PM / block_m);
  TORCH_CHECK(pos.numel() == n);
  fi::launch_moe_align(topi.data_ptr<int>(), sorted_ids.data_ptr<int>(),
                       expert_ids.data_ptr<int>(), n_valid.data_ptr<int>(),
                       pos.data_ptr<int>(), n, (int)topk, (int)e_start,
                       (int)e_end, (int)block_m, PM, current_stream());
}

void moe_router_topk(at::Tensor logits, at::Tensor topv, at::Tensor topi,
                     int64_t k, bool renorm) {
  TORCH_CHECK(logits.is_cuda() && logits.scalar_type() == at::kFloat &&
              logits.is_contiguous());
  TORCH_CHECK(topv.scalar_type() == at::kFloat &&
              topi.scalar_type() == at::kInt);
  const int T = logits.size(0);
  const int E = logits.size(1);
  TORCH_CHECK(E <= 128 && k <= E);
  TORCH_CHECK(topv.numel() == T * k && topi.numel() == T * k);
  fi::launch_moe_router_topk(logits.data_ptr<float>(),
                             topv.data_ptr<float>(), topi.data_ptr<int>(),
                             T, E, (int)k, renorm, current_stream());
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rms_norm", &rms_norm, "RMSNorm (bf16, fp32 accum)");
  m.def("fused_add_rms_norm", &fused_add_rms_norm,
        "in-place residual add + RMSNorm");
  m.def("silu_and_mul", &silu_and_mul, "fused SiLU-mul");
  m.def("rms_norm_fp8", &rms_norm_fp8, "RMSNorm with fused fp8 row quant");
  m.def("fused_add_rms_norm_fp8", &fused_add_rms_norm_fp8,
        "residual add + RMSNorm with fused fp8 row quant");
  m.def("silu_and_mul_fp8", &silu_and_mul_fp8,
        "fused SiLU-mul with fp8 row quant");
  m.def("quant_fp8_rows", &quant_fp8_rows, "per-row dynamic fp8 quant");
  m.def("rope_qk_norm", &rope_qk_norm,
        "fused per-head qk RMSNorm + NeoX RoPE (in-place)");
  m.def("reshape_and_cache", &reshape_and_cache, "scatter K/V into paged cache");
  m.def("gather_kv_blocks", &gather_kv_blocks, "pack KV blocks for PD send");
  m.def("scatter_kv_blocks", &scatter_kv_blocks, "unpack KV blocks from PD recv");
  m.def("paged_attention_decode", &paged_attention_decode,
        "paged decode attention");
  m.def("prefill_attention", &prefill_attention, "varlen causal MFMA prefill");
  m.def("prefill_attention_paged", &prefill_attention_paged,
        "varlen causal MFMA prefill over the paged cache (context attention)");
  m.def("moe_gemm", &moe_gemm,
        "grouped MFMA GEMM over block-aligned expert segments "
        "(gate_up=true fuses the SwiGLU epilogue)");
  m.def("moe_gemm_fp8", &moe_gemm_fp8,
        "grouped fp8 (e4m3) MFMA GEMM with per-row/per-channel dequant "
        "epilogue (gate_up=true fuses SwiGLU)");
  m.def("moe_combine", &moe_combine,
        "weighted top-k combine of expert outputs (deterministic)");
  m.def("moe_align", &moe_align,
        "single-kernel block alignment of expert assignments "
        "(vLLM moe_align_block_size analog)");
  m.def("moe_router_topk", &moe_router_topk,
        "fused router tail: softmax + top-k + optional renormalize");
}
