// Varlen causal prefill attention (flash-style, MFMA) for CDNA4 (gfx950) — v2.
//
// out[t, h, :] = softmax(Q K^T * scale, causal) V, GQA, bf16 in / fp32 accum.
//
// v2 structure (the 8-wave ladder of cdna_hip_programming.md §B):
//  * workgroup = 8 waves; wave w owns q rows [row0 + 32w, +32) of one
//    (seq, q-head) — a 256-row Q block per workgroup.
//  * KV tiles of 64 tokens are staged ONCE per workgroup into shared LDS
//    (cooperative coalesced loads, barrier-synced): an 8x cut in global
//    K/V traffic vs per-wave reads (v1's top cost: 26% of bench GPU time).
//  * K tile [64][256B] is XOR-swizzled (byte ^= (row&15)<<4) so the QK^T
//    B-fragment ds_read_b128 is bank-conflict-free (guide T2/G4);
//    V is staged TRANSPOSED ([128][128B] rows, byte ^= (row&7)<<4) so the
//    PV B-fragment is a contiguous swizzled b128 read.
//  * QK^T and PV both mfma_f32_16x16x32_bf16; P goes through per-wave
//    swizzled LDS (bf16) to become the PV A-fragment.
//  * online softmax per 16-row fragment block; C-fragment mapping:
//    col = lane&15, row = (lane>>4)*4 + reg.
//
// Capability parity: the paged-attention prefill the reference delegates to
// vLLM (SURVEY.md §2.3 "Paged-attention prefill kernel").

#include "common.h"

namespace fi {

typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(4))) float floatx4;

constexpr int kWaves = 8;
constexpr int kQPerWave = 32;                 // q rows per wave (2 MFMA blocks)
constexpr int kQBlock = kWaves * kQPerWave;   // 256 q rows per workgroup
constexpr int kKVTile = 64;                   // kv tokens per LDS tile
constexpr float kPNegInf = -1e30f;

// byte-address XOR swizzles, bijective within a ROWB-byte row (guide T2/G4):
// spread a 16-lane group's distinct-row b128 reads over ROWB/16 bank slots.
template <int ROWB>
FI_DEV int swz(int row, int byte_in_row) {
  constexpr int kMask = (ROWB / 16 > 16 ? 16 : ROWB / 16) - 1;
  return row * ROWB + (byte_in_row ^ ((row & kMask) << 4));
}

template <int D>
__global__ __launch_bounds__(kWaves * kWaveSize) void prefill_attn_kernel(
    u16* __restrict__ out,        // [T, Hq, D]
    const u16* __restrict__ q,    // [T] rows, stride q_stride
    const u16* __restrict__ k,    // [T] rows, stride k_stride
    const u16* __restrict__ v,    // [T] rows, stride v_stride
    const int* __restrict__ tile_seq,    // [ntiles] sequence index
    const int* __restrict__ tile_row0,   // [ntiles] first q row (within seq)
    const int* __restrict__ cu_seqlens,  // [nseqs+1]
    const int64_t q_stride, const int64_t k_stride, const int64_t v_stride,
    const int num_q_heads, const int num_kv_heads, const float scale) {
  constexpr int KB = D / 32;   // QK^T k-chunks over the head dim
  constexpr int CB = D / 16;   // PV output col blocks
  constexpr int kKRowB = D * 2;        // K row bytes (D=128 -> 256)
  constexpr int kVTRowB = kKVTile * 2; // V^T row bytes (64 kv -> 128)

  const int head = blockIdx.y;
  const int kv_head = head / (num_q_heads / num_kv_heads);
  const int tid = threadIdx.x;
  const int wave = tid / kWaveSize;
  const int lane = tid % kWaveSize;
  const int col = lane & 15;
  const int hi = lane >> 4;

  const int seq = tile_seq[blockIdx.x];
  const int seq_start = cu_seqlens[seq];
  const int seq_len = cu_seqlens[seq + 1] - seq_start;
  const int wg_row0 = tile_row0[blockIdx.x];
  const int row0 = wg_row0 + wave * kQPerWave;      // this wave's first q row
  const bool active = row0 < seq_len;

  // LDS: K tile + V^T tile (workgroup-shared) + P tiles (per-wave)
  __shared__ u16 k_lds[kKVTile * D];            // swizzled rows of 256 B
  __shared__ u16 vt_lds[D * kKVTile];           // transposed, swizzled 128-B rows
  __shared__ u16 p_lds[kWaves][kQPerWave * kKVTile];  // swizzled 128-B rows

  // ---- Q fragments: a_q[rb][kb], lane holds Q[row0+rb*16+col][kb*32+hi*8..]
  short8 a_q[2][KB];
  if (active) {
#pragma unroll
    for (int rb = 0; rb < 2; ++rb) {
      const int qr = min(row0 + rb * 16 + col, seq_len - 1);
      const u16* qrow = q + (seq_start + qr) * q_stride +
                        static_cast<int64_t>(head) * D;
#pragma unroll
      for (int kb = 0; kb < KB; ++kb)
        a_q[rb][kb] = *reinterpret_cast<const short8*>(qrow + kb * 32 + hi * 8);
    }
  }

  float m[2][4], l[2][4];
  floatx4 o_acc[2][CB];
#pragma unroll
  for (int rb = 0; rb < 2; ++rb) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      m[rb][r] = kPNegInf;
      l[rb][r] = 0.f;
    }
#pragma unroll
    for (int cb = 0; cb < CB; ++cb) o_acc[rb][cb] = {0.f, 0.f, 0.f, 0.f};
  }

  // kv range: the whole workgroup iterates to its max causal row
  const int wg_q_max = min(wg_row0 + kQBlock, seq_len) - 1;
  const int num_kv_tiles = wg_q_max / kKVTile + 1;
  const int my_q_max = min(row0 + kQPerWave, seq_len) - 1;  // per-wave

  for (int t = 0; t < num_kv_tiles; ++t) {
    const int kv0 = t * kKVTile;

    // ---- cooperative staging: K [64][D] swizzled; V^T [D][64] swizzled ----
    {
      // one unit = 16 consecutive elements of one K/V row
      constexpr int kChunks = D / 16;  // 16-elem chunks per row
      for (int u = tid; u < kKVTile * kChunks; u += kWaves * kWaveSize) {
        const int kv_r = u / kChunks;
        const int c16 = u % kChunks;
        const int src = min(kv0 + kv_r, seq_len - 1);
        const u16* krow = k + (seq_start + src) * k_stride +
                          static_cast<int64_t>(kv_head) * D + c16 * 16;
        const u16* vrow = v + (seq_start + src) * v_stride +
                          static_cast<int64_t>(kv_head) * D + c16 * 16;
        bf16x8 k0 = *reinterpret_cast<const bf16x8*>(krow);
        bf16x8 k1 = *reinterpret_cast<const bf16x8*>(krow + 8);
        bf16x8 v0 = *reinterpret_cast<const bf16x8*>(vrow);
        bf16x8 v1 = *reinterpret_cast<const bf16x8*>(vrow + 8);
        char* kbase = reinterpret_cast<char*>(k_lds);
        *reinterpret_cast<bf16x8*>(kbase + swz<kKRowB>(kv_r, c16 * 32)) = k0;
        *reinterpret_cast<bf16x8*>(kbase + swz<kKRowB>(kv_r, c16 * 32 + 16)) = k1;
        // V transpose: element (kv_r, d) -> vt row d, col kv_r
        char* vbase = reinterpret_cast<char*>(vt_lds);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const int d0 = c16 * 16 + j;
          const int d1 = d0 + 8;
          *reinterpret_cast<u16*>(
              vbase + swz<kVTRowB>(d0, kv_r * 2)) = v0.h[j];
          *reinterpret_cast<u16*>(
              vbase + swz<kVTRowB>(d1, kv_r * 2)) = v1.h[j];
        }
      }
    }
    __syncthreads();

    const bool compute = active && kv0 <= my_q_max;
    if (compute) {
      const char* kbase = reinterpret_cast<const char*>(k_lds);
      const char* vbase = reinterpret_cast<const char*>(vt_lds);
      char* pbase = reinterpret_cast<char*>(p_lds[wave]);

#pragma unroll
      for (int rb = 0; rb < 2; ++rb) {
        // ---- QK^T: S[16 x 64] ----
        floatx4 s_acc[4];
#pragma unroll
        for (int cb4 = 0; cb4 < 4; ++cb4) {
          s_acc[cb4] = {0.f, 0.f, 0.f, 0.f};
          const int krow_idx = cb4 * 16 + col;  // kv token within tile
#pragma unroll
          for (int kb = 0; kb < KB; ++kb) {
            const short8 b_k = *reinterpret_cast<const short8*>(
                kbase + swz<kKRowB>(krow_idx, kb * 32 * 2 + hi * 16));
            s_acc[cb4] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a_q[rb][kb], b_k, s_acc[cb4], 0, 0, 0);
          }
        }

        // ---- mask + online softmax over the 16x64 block ----
        float p[4][4];
        const int q_base = row0 + rb * 16 + hi * 4;
#pragma unroll
        for (int cb4 = 0; cb4 < 4; ++cb4) {
          const int kv_pos = kv0 + cb4 * 16 + col;
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const int q_pos = q_base + r;
            float sv = s_acc[cb4][r] * scale;
            if (kv_pos > q_pos || q_pos >= seq_len || kv_pos >= seq_len)
              sv = kPNegInf;
            p[cb4][r] = sv;
          }
        }
        float alpha[4];
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          float rm = fmaxf(fmaxf(p[0][r], p[1][r]), fmaxf(p[2][r], p[3][r]));
#pragma unroll
          for (int off = 1; off < 16; off <<= 1)
            rm = fmaxf(rm, __shfl_xor(rm, off, 64));
          const float m_new = fmaxf(m[rb][r], rm);
          if (m_new <= kPNegInf) {  // row fully masked so far
            alpha[r] = 0.f;
#pragma unroll
            for (int cb4 = 0; cb4 < 4; ++cb4) p[cb4][r] = 0.f;
            continue;
          }
          alpha[r] = (m[rb][r] <= kPNegInf) ? 0.f : __expf(m[rb][r] - m_new);
          m[rb][r] = m_new;
          float rs = 0.f;
#pragma unroll
          for (int cb4 = 0; cb4 < 4; ++cb4) {
            p[cb4][r] =
                (p[cb4][r] <= kPNegInf) ? 0.f : __expf(p[cb4][r] - m_new);
            rs += p[cb4][r];
          }
#pragma unroll
          for (int off = 1; off < 16; off <<= 1) rs += __shfl_xor(rs, off, 64);
          l[rb][r] = l[rb][r] * alpha[r] + rs;
        }

        // ---- P -> per-wave LDS (bf16, swizzled 128-B rows) ----
#pragma unroll
        for (int cb4 = 0; cb4 < 4; ++cb4)
#pragma unroll
          for (int r = 0; r < 4; ++r)
            *reinterpret_cast<u16*>(
                pbase + swz<kVTRowB>(rb * 16 + hi * 4 + r,
                                        (cb4 * 16 + col) * 2)) =
                f32_to_bf16(p[cb4][r]);

        // ---- PV: O[16 x D] += P[16 x 64] V[64 x D] ----
#pragma unroll
        for (int kc = 0; kc < 2; ++kc) {  // kv 64 = 2 MFMA k-depths
          const short8 a_p = *reinterpret_cast<const short8*>(
              pbase + swz<kVTRowB>(rb * 16 + col, kc * 64 + hi * 16));
#pragma unroll
          for (int cb = 0; cb < CB; ++cb) {
            const short8 b_v = *reinterpret_cast<const short8*>(
                vbase + swz<kVTRowB>(cb * 16 + col, kc * 64 + hi * 16));
            floatx4 prev = o_acc[rb][cb];
            if (kc == 0) {
#pragma unroll
              for (int r = 0; r < 4; ++r) prev[r] *= alpha[r];
            }
            o_acc[rb][cb] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a_p, b_v, prev, 0, 0, 0);
          }
        }
      }
    }
    __syncthreads();
  }

  // ---- epilogue: normalize and store ----
  if (active) {
#pragma unroll
    for (int rb = 0; rb < 2; ++rb) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int q_pos = row0 + rb * 16 + hi * 4 + r;
        if (q_pos >= seq_len) continue;
        const float inv = (l[rb][r] > 0.f) ? 1.f / l[rb][r] : 0.f;
        u16* orow = out + (static_cast<int64_t>(seq_start + q_pos) *
                               num_q_heads + head) * D;
#pragma unroll
        for (int cb = 0; cb < CB; ++cb)
          orow[cb * 16 + col] = f32_to_bf16(o_acc[rb][cb][r] * inv);
      }
    }
  }
}

void launch_prefill_attn(u16* out, const u16* q, const u16* k, const u16* v,
                         const int* tile_seq, const int* tile_row0,
                         const int* cu_seqlens, int ntiles, int64_t q_stride,
                         int64_t k_stride, int64_t v_stride, int num_q_heads,
                         int num_kv_heads, int head_dim, float scale,
                         hipStream_t stream) {
  dim3 grid(ntiles, num_q_heads), block(kWaves * kWaveSize);
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
