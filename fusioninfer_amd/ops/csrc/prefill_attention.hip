// Varlen causal prefill attention (flash-style, MFMA) for CDNA4 (gfx950).
//
// out[t, h, :] = softmax(Q K^T * scale, causal) V over each sequence's own
// rows, GQA (G q-heads per kv-head), bf16 in / fp32 accumulate.
//
// v1 structure (correctness-first; the optimization ladder in
// cdna_hip_programming.md §B raises this later):
//  * workgroup = 4 independent waves; each wave owns a 16-row Q tile of one
//    (seq, q-head) and runs its own online-softmax loop over 32-token KV
//    tiles — no cross-wave barriers in the main loop.
//  * QK^T: mfma_f32_16x16x32_bf16. With row-major Q and K, BOTH fragments
//    are contiguous bf16x8 loads: A[i][k] = Q[row i][k-chunk], and
//    B[k][j] = K^T[k][j] = K[row j][k-chunk] (the "B^T input" form).
//  * P goes through LDS (bf16) so the PV A-fragment is a ds_read_b128;
//    V^T is staged per-wave into LDS so the PV B-fragment is contiguous.
//  * online softmax per 16x32 S-tile; C-fragment mapping (guide §3):
//    col = lane&15, row = (lane>>4)*4 + reg.
//
// Capability parity: the paged-attention prefill the reference delegates to
// vLLM (SURVEY.md §2.3 "Paged-attention prefill kernel").

#include "common.h"

namespace fi {

typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(4))) float floatx4;

constexpr int kQTile = 16;    // q rows per wave
constexpr int kKVTile = 32;   // kv tokens per tile (= one MFMA K depth)
constexpr int kPFWaves = 4;
constexpr float kPNegInf = -1e30f;

template <int D>
__global__ __launch_bounds__(kPFWaves * kWaveSize) void prefill_attn_kernel(
    u16* __restrict__ out,        // [T, Hq, D]
    const u16* __restrict__ q,    // [T] rows, stride q_stride
    const u16* __restrict__ k,    // [T] rows, stride k_stride
    const u16* __restrict__ v,    // [T] rows, stride v_stride
    const int* __restrict__ tile_seq,    // [ntiles] sequence index
    const int* __restrict__ tile_row0,   // [ntiles] first q row (within seq)
    const int* __restrict__ cu_seqlens,  // [nseqs+1]
    const int64_t q_stride, const int64_t k_stride, const int64_t v_stride,
    const int num_q_heads, const int num_kv_heads, const float scale) {
  constexpr int KB = D / 32;   // MFMA k-chunks over the head dim
  constexpr int CB = D / 16;   // output col blocks

  const int head = blockIdx.y;
  const int kv_head = head / (num_q_heads / num_kv_heads);
  const int wave = threadIdx.x / kWaveSize;
  const int lane = threadIdx.x % kWaveSize;
  const int col = lane & 15;       // fragment col / B row-token
  const int hi = lane >> 4;        // fragment 8-chunk index (0..3)

  const int seq = tile_seq[blockIdx.x];
  const int seq_start = cu_seqlens[seq];
  const int seq_len = cu_seqlens[seq + 1] - seq_start;
  const int row0 = tile_row0[blockIdx.x] + wave * kQTile;  // within seq
  if (row0 >= seq_len) return;

  // LDS: per-wave V^T tile + per-wave P tile
  __shared__ u16 vt_lds[kPFWaves][D][kKVTile];        // V^T (dim-major)
  __shared__ u16 p_lds[kPFWaves][kQTile][kKVTile];

  // ---- load Q fragments (A): lane holds Q[row0+col][kb*32 + hi*8 .. +8]
  short8 a_q[KB];
  {
    const int qr = min(row0 + col, seq_len - 1);  // clamp; masked later
    const u16* qrow = q + (seq_start + qr) * q_stride +
                      static_cast<int64_t>(head) * D;
#pragma unroll
    for (int kb = 0; kb < KB; ++kb)
      a_q[kb] = *reinterpret_cast<const short8*>(qrow + kb * 32 + hi * 8);
  }

  float m[4], l[4];
  floatx4 o_acc[CB];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    m[r] = kPNegInf;
    l[r] = 0.f;
  }
#pragma unroll
  for (int cb = 0; cb < CB; ++cb) o_acc[cb] = {0.f, 0.f, 0.f, 0.f};

  const int q_max = min(row0 + kQTile, seq_len) - 1;   // last valid q row
  const int num_kv_tiles = q_max / kKVTile + 1;

  for (int t = 0; t < num_kv_tiles; ++t) {
    const int kv0 = t * kKVTile;
    const int kv_valid = min(seq_len - kv0, kKVTile);

    // ---- QK^T: S[16 x 32] = Q[16 x D] K^T ----
    floatx4 s_acc[2];
    s_acc[0] = {0.f, 0.f, 0.f, 0.f};
    s_acc[1] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int ch = 0; ch < 2; ++ch) {
      const int kt = min(kv0 + ch * 16 + col, seq_len - 1);
      const u16* krow = k + (seq_start + kt) * k_stride +
                        static_cast<int64_t>(kv_head) * D;
#pragma unroll
      for (int kb = 0; kb < KB; ++kb) {
        const short8 b_k =
            *reinterpret_cast<const short8*>(krow + kb * 32 + hi * 8);
        s_acc[ch] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_q[kb], b_k,
                                                            s_acc[ch], 0, 0, 0);
      }
    }

    // ---- stage V^T into LDS (each lane: one V row, 8 dims per store loop)
    {
      // 64 lanes cover 32 rows x (D/8 col-chunks per half): lane maps to
      // row = lane % 32, chunk = lane / 32, strided by 2 chunks.
      const int vrow = lane & 31;
      const int kt = min(kv0 + vrow, seq_len - 1);
      const u16* vr = v + (seq_start + kt) * v_stride +
                      static_cast<int64_t>(kv_head) * D;
      for (int c8 = lane >> 5; c8 < D / 8; c8 += 2) {
        bf16x8 vv = *reinterpret_cast<const bf16x8*>(vr + c8 * 8);
#pragma unroll
        for (int j = 0; j < 8; ++j) vt_lds[wave][c8 * 8 + j][vrow] = vv.h[j];
      }
    }

    // ---- mask + online softmax on the 16x32 S tile ----
    float p[2][4];
#pragma unroll
    for (int ch = 0; ch < 2; ++ch) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int q_pos = row0 + hi * 4 + r;
        const int kv_pos = kv0 + ch * 16 + col;
        float sv = s_acc[ch][r] * scale;
        if (kv_pos > q_pos || q_pos >= seq_len || kv_pos >= seq_len)
          sv = kPNegInf;
        p[ch][r] = sv;
      }
    }
    float alpha[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float rm = fmaxf(p[0][r], p[1][r]);
#pragma unroll
      for (int off = 1; off < 16; off <<= 1)
        rm = fmaxf(rm, __shfl_xor(rm, off, 64));
      const float m_new = fmaxf(m[r], rm);
      alpha[r] = (m[r] <= kPNegInf) ? 0.f : __expf(m[r] - m_new);
      if (m_new <= kPNegInf) {  // fully-masked row so far
        alpha[r] = 0.f;
        m[r] = m_new;
        p[0][r] = 0.f;
        p[1][r] = 0.f;
        continue;
      }
      m[r] = m_new;
      float rs = 0.f;
#pragma unroll
      for (int ch = 0; ch < 2; ++ch) {
        p[ch][r] = (p[ch][r] <= kPNegInf) ? 0.f : __expf(p[ch][r] - m_new);
        rs += p[ch][r];
      }
#pragma unroll
      for (int off = 1; off < 16; off <<= 1) rs += __shfl_xor(rs, off, 64);
      l[r] = l[r] * alpha[r] + rs;
    }

    // ---- P -> LDS (bf16) ----
#pragma unroll
    for (int ch = 0; ch < 2; ++ch)
#pragma unroll
      for (int r = 0; r < 4; ++r)
        p_lds[wave][hi * 4 + r][ch * 16 + col] = f32_to_bf16(p[ch][r]);

    // zero the tail of V^T's kv rows so garbage never enters PV
    if (kv_valid < kKVTile) {
      const int vrow = lane & 31;
      if (vrow >= kv_valid)
        for (int c8 = lane >> 5; c8 < D / 8; c8 += 2)
#pragma unroll
          for (int j = 0; j < 8; ++j) vt_lds[wave][c8 * 8 + j][vrow] = 0;
    }

    // ---- PV: O[16 x D] += P[16 x 32] V[32 x D] ----
    // A-frag: lane holds P[col][hi*8 .. +8] (contiguous in p_lds)
    const short8 a_p = *reinterpret_cast<const short8*>(&p_lds[wave][col][hi * 8]);
#pragma unroll
    for (int cb = 0; cb < CB; ++cb) {
      // B-frag: lane holds V^T[cb*16 + col][hi*8 .. +8] (contiguous)
      const short8 b_v =
          *reinterpret_cast<const short8*>(&vt_lds[wave][cb * 16 + col][hi * 8]);
      floatx4 prev = o_acc[cb];
#pragma unroll
      for (int r = 0; r < 4; ++r) prev[r] *= alpha[r];
      o_acc[cb] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_p, b_v, prev, 0, 0, 0);
    }
  }

  // ---- epilogue: normalize and store ----
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int q_pos = row0 + hi * 4 + r;
    if (q_pos >= seq_len) continue;
    const float inv = (l[r] > 0.f) ? 1.f / l[r] : 0.f;
    u16* orow = out + (static_cast<int64_t>(seq_start + q_pos) * num_q_heads +
                       head) * D;
#pragma unroll
    for (int cb = 0; cb < CB; ++cb)
      orow[cb * 16 + col] = f32_to_bf16(o_acc[cb][r] * inv);
  }
}

void launch_prefill_attn(u16* out, const u16* q, const u16* k, const u16* v,
                         const int* tile_seq, const int* tile_row0,
                         const int* cu_seqlens, int ntiles, int64_t q_stride,
                         int64_t k_stride, int64_t v_stride, int num_q_heads,
                         int num_kv_heads, int head_dim, float scale,
                         hipStream_t stream) {
  dim3 grid(ntiles, num_q_heads), block(kPFWaves * kWaveSize);
  if (head_dim == 128) {
    hipLaunchKernelGGL((prefill_attn_kernel<128>), grid, block, 0, stream, out,
                       q, k, v, tile_seq, tile_row0, cu_seqlens, q_stride,
                       k_stride, v_stride, num_q_heads, num_kv_heads, scale);
  } else if (head_dim == 64) {
    hipLaunchKernelGGL((prefill_attn_kernel<64>), grid, block, 0, stream, out,
                       q, k, v, tile_seq, tile_row0, cu_seqlens, q_stride,
                       k_stride, v_stride, num_q_heads, num_kv_heads, scale);
  } else {
    abort();
  }
}

}  // namespace fi
