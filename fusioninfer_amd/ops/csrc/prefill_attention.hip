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

// PAGED: K/V come from the paged cache ([B, Hk, 16, D]) via block_tables,
// and q rows are only the NEW tokens of each sequence (context attention —
// prefix-cache hits / chunked prefill skip cached tokens).
template <int D, bool PAGED>
__global__ __launch_bounds__(kWaves * kWaveSize) void prefill_attn_kernel(
    u16* __restrict__ out,        // [T, Hq, D]
    const u16* __restrict__ q,    // [T] rows, stride q_stride
    const u16* __restrict__ k,    // dense rows OR k_cache when PAGED
    const u16* __restrict__ v,    // dense rows OR v_cache when PAGED
    const int* __restrict__ tile_seq,    // [ntiles] sequence index
    const int* __restrict__ tile_row0,   // [ntiles] first NEW q row
    const int* __restrict__ cu_seqlens,  // [nseqs+1] over NEW tokens
    const int* __restrict__ block_tables,  // [S, max_blocks] (PAGED)
    const int* __restrict__ seq_lens_k,    // [S] total ctx len (PAGED)
    const int max_blocks,
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
  const int seq_len = cu_seqlens[seq + 1] - seq_start;  // NEW q rows
  const int k_len = PAGED ? seq_lens_k[seq] : seq_len;  // total kv rows
  const int ctx_start = k_len - seq_len;  // cached tokens before q row 0
  const int wg_row0 = tile_row0[blockIdx.x];
  const int row0 = wg_row0 + wave * kQPerWave;      // this wave's first q row
  const bool active = row0 < seq_len;
  const int* bt_row = PAGED ? block_tables + static_cast<int64_t>(seq) * max_blocks
                            : nullptr;
  // element offset of kv row r (absolute position) in the k/v source
  auto kv_off = [&](int r, int64_t dense_stride) -> int64_t {
    if (PAGED) {
      const int blk = bt_row[r >> 4];
      return ((static_cast<int64_t>(blk) * num_kv_heads + kv_head) * 16 +
              (r & 15)) * D;
    }
    return (seq_start + r) * dense_stride + static_cast<int64_t>(kv_head) * D;
  };

  // LDS: double-buffered K (glds target) + V^T tile + P tiles (per-wave).
  // ONE __shared__ object: a second one makes hipcc drain vmcnt(0) before
  // every ds_read beside a glds pipeline (guide §5 ".s-level traps" (a)).
  __shared__ u16 smem[2 * kKVTile * D + D * kKVTile +
                      kWaves * kQPerWave * kKVTile];
  u16* k_lds0 = smem;                            // swizzled K rows, buffer 0
  u16* k_lds1 = smem + kKVTile * D;              // buffer 1
  u16* vt_lds = smem + 2 * kKVTile * D;          // transposed V, swizzled rows
  u16* p_lds = vt_lds + D * kKVTile + wave * (kQPerWave * kKVTile);

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

  // kv range: the whole workgroup iterates to its max causal ABSOLUTE row
  const int wg_q_max = ctx_start + min(wg_row0 + kQBlock, seq_len) - 1;
  const int num_kv_tiles = wg_q_max / kKVTile + 1;
  const int my_q_max = ctx_start + min(row0 + kQPerWave, seq_len) - 1;

  // ---- staging helpers -------------------------------------------------
  // K tile -> LDS by global_load_lds (direct DMA, no VGPR round trip).
  // glds writes lane-linear (base + lane*16), so the T2 swizzle moves to
  // the per-lane SOURCE address (guide §5.4 rule 21).
  constexpr int kKTileBytes = kKVTile * kKRowB;
  constexpr int kGldsPerWave = kKTileBytes / (kWaves * kWaveSize * 16);
  auto stage_k_glds = [&](int t, u16* kbuf) {
    const int kv0 = t * kKVTile;
#pragma unroll
    for (int i = 0; i < kGldsPerWave; ++i) {
      const int base_off =
          wave * (kKTileBytes / kWaves) + i * (kWaveSize * 16);
      const int X = base_off + lane * 16;
      const int row = X / kKRowB;
      const int sbyte = X % kKRowB;
      constexpr int kMask = (kKRowB / 16 > 16 ? 16 : kKRowB / 16) - 1;
      const int byte = sbyte ^ ((row & kMask) << 4);
      const int src_row = min(kv0 + row, k_len - 1);
      const u16* gsrc = k + kv_off(src_row, k_stride) + byte / 2;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)gsrc,
          (__attribute__((address_space(3))) void*)(
              reinterpret_cast<char*>(kbuf) + base_off),
          16, 0, 0);
    }
  };
  // V: per-thread register load of one 16-elem row chunk (issued early so
  // HBM latency hides under MFMA — guide T14), transposed into LDS later.
  constexpr int kChunks = D / 16;
  const int v_kv = tid % kKVTile;          // this thread's kv row
  const int v_c16 = tid / kKVTile;         // 16-elem chunk (D=128: 0..7)
  bf16x8 vreg0, vreg1;
  auto vload = [&](int t) {
    if (v_c16 >= kChunks) return;  // D=64: only 4 chunks per row
    const int src = min(t * kKVTile + v_kv, k_len - 1);
    const u16* vrow = v + kv_off(src, v_stride) + v_c16 * 16;
    vreg0 = *reinterpret_cast<const bf16x8*>(vrow);
    vreg1 = *reinterpret_cast<const bf16x8*>(vrow + 8);
  };
  auto vwrite = [&]() {
    if (v_c16 >= kChunks) return;
    char* vbase = reinterpret_cast<char*>(vt_lds);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      *reinterpret_cast<u16*>(
          vbase + swz<kVTRowB>(v_c16 * 16 + j, v_kv * 2)) = vreg0.h[j];
      *reinterpret_cast<u16*>(
          vbase + swz<kVTRowB>(v_c16 * 16 + j + 8, v_kv * 2)) = vreg1.h[j];
    }
  };

  // ---- software-pipelined main loop -------------------------------------
  // invariant at loop top: K(t) resident in kbuf(t&1) (glds issued and
  // drained by the previous barrier), V(t) in registers.
  stage_k_glds(0, k_lds0);
  vload(0);
  __syncthreads();  // drains the glds (vmcnt 0) and publishes K(0)

  for (int t = 0; t < num_kv_tiles; ++t) {
    const int kv0 = t * kKVTile;
    u16* kbuf = (t & 1) ? k_lds1 : k_lds0;
    u16* kbuf_next = (t & 1) ? k_lds0 : k_lds1;

    vwrite();  // V(t) -> vt_lds (read after the mid barrier)
    if (t + 1 < num_kv_tiles)
      stage_k_glds(t + 1, kbuf_next);  // flies under QK^T(t)

    const bool compute = active && kv0 <= my_q_max;
    if (compute) {
      const char* kbase = reinterpret_cast<const char*>(kbuf);
      char* pbase = reinterpret_cast<char*>(p_lds);

#pragma unroll
      for (int rb = 0; rb < 2; ++rb) {
        // ---- QK^T: S[16 x 64] ----
        floatx4 s_acc[4];
        __builtin_amdgcn_s_setprio(1);
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
        __builtin_amdgcn_s_setprio(0);

        // ---- mask + online softmax over the 16x64 block (in s_acc) ----
        const int q_base = row0 + rb * 16 + hi * 4;
#pragma unroll
        for (int cb4 = 0; cb4 < 4; ++cb4) {
          const int kv_pos = kv0 + cb4 * 16 + col;  // absolute position
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const int q_pos = ctx_start + q_base + r;  // absolute position
            float sv = s_acc[cb4][r] * scale;
            if (kv_pos > q_pos || q_base + r >= seq_len || kv_pos >= k_len)
              sv = kPNegInf;
            s_acc[cb4][r] = sv;
          }
        }
        // o_acc is rescaled HERE (phase 1) so nothing but P crosses the
        // barrier into the PV phase
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          float rm = fmaxf(fmaxf(s_acc[0][r], s_acc[1][r]),
                           fmaxf(s_acc[2][r], s_acc[3][r]));
#pragma unroll
          for (int off = 1; off < 16; off <<= 1)
            rm = fmaxf(rm, __shfl_xor(rm, off, 64));
          const float m_new = fmaxf(m[rb][r], rm);
          float alpha;
          if (m_new <= kPNegInf) {  // row fully masked so far
            alpha = 0.f;
          } else {
            alpha = (m[rb][r] <= kPNegInf) ? 0.f : __expf(m[rb][r] - m_new);
            m[rb][r] = m_new;
          }
#pragma unroll
          for (int cb = 0; cb < CB; ++cb) o_acc[rb][cb][r] *= alpha;
          float rs = 0.f;
#pragma unroll
          for (int cb4 = 0; cb4 < 4; ++cb4) {
            const float pv = (m_new <= kPNegInf || s_acc[cb4][r] <= kPNegInf)
                                 ? 0.f
                                 : __expf(s_acc[cb4][r] - m_new);
            rs += pv;
            *reinterpret_cast<u16*>(
                pbase + swz<kVTRowB>(rb * 16 + hi * 4 + r,
                                     (cb4 * 16 + col) * 2)) = f32_to_bf16(pv);
          }
#pragma unroll
          for (int off = 1; off < 16; off <<= 1) rs += __shfl_xor(rs, off, 64);
          l[rb][r] = l[rb][r] * alpha + rs;
        }
      }
    }

    // publish vt_lds(t) to every wave; also drains the K(t+1) glds, whose
    // latency QK^T just covered
    __syncthreads();
    if (t + 1 < num_kv_tiles)
      vload(t + 1);  // flies under PV(t); consumed by vwrite next iteration

    if (compute) {
      const char* vbase = reinterpret_cast<const char*>(vt_lds);
      char* pbase = reinterpret_cast<char*>(p_lds);
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int rb = 0; rb < 2; ++rb) {
        // ---- PV: O[16 x D] += P[16 x 64] V[64 x D] ----
#pragma unroll
        for (int kc = 0; kc < 2; ++kc) {  // kv 64 = 2 MFMA k-depths
          const short8 a_p = *reinterpret_cast<const short8*>(
              pbase + swz<kVTRowB>(rb * 16 + col, kc * 64 + hi * 16));
#pragma unroll
          for (int cb = 0; cb < CB; ++cb) {
            const short8 b_v = *reinterpret_cast<const short8*>(
                vbase + swz<kVTRowB>(cb * 16 + col, kc * 64 + hi * 16));
            o_acc[rb][cb] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a_p, b_v, o_acc[rb][cb], 0, 0, 0);
          }
        }
      }
      __builtin_amdgcn_s_setprio(0);
    }
    // PV(t) reads of vt_lds done before vwrite(t+1) overwrites it
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
                         const int* cu_seqlens, const int* block_tables,
                         const int* seq_lens_k, int max_blocks, int ntiles,
                         int64_t q_stride, int64_t k_stride, int64_t v_stride,
                         int num_q_heads, int num_kv_heads, int head_dim,
                         float scale, hipStream_t stream) {
  dim3 grid(ntiles, num_q_heads), block(kWaves * kWaveSize);
  const bool paged = block_tables != nullptr;
#define FI_PF_LAUNCH(DD, PP)                                                  \
  hipLaunchKernelGGL((prefill_attn_kernel<DD, PP>), grid, block, 0, stream,   \
                     out, q, k, v, tile_seq, tile_row0, cu_seqlens,           \
                     block_tables, seq_lens_k, max_blocks, q_stride,          \
                     k_stride, v_stride, num_q_heads, num_kv_heads, scale)
  if (head_dim == 128) {
    if (paged) FI_PF_LAUNCH(128, true); else FI_PF_LAUNCH(128, false);
  } else if (head_dim == 64) {
    if (paged) FI_PF_LAUNCH(64, true); else FI_PF_LAUNCH(64, false);
  } else {
    abort();
  }
#undef FI_PF_LAUNCH
}

}  // namespace fi
