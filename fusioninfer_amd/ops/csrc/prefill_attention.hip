// Varlen causal prefill / context attention (flash-style, MFMA) — v3.
//
// out[t, h, :] = softmax(Q K^T * scale, causal) V, GQA, bf16 in / fp32
// accumulate. PAGED variant reads K/V from the paged cache via block tables
// (q rows are only each sequence's NEW tokens — prefix-cache hits and
// chunked prefill skip cached tokens).
//
// v3 structure (the 8-wave 32x32 swapped-operand ladder of
// cdna_hip_programming.md §B):
//  * workgroup = 8 waves; wave w owns q rows [row0 + 32w, +32); 256 q rows
//    per workgroup; KV tiles of 64 staged once per workgroup.
//  * K by global_load_lds (double-buffered, T2 swizzle on the SOURCE
//    address); V register-staged early (T14) and written TRANSPOSED
//    (V^T, swizzled rows) so the PV B-fragment is a clean ds_read_b128.
//  * SWAPPED QK^T: S^T = mfma_f32_32x32x16_bf16(A=K, B=Q) puts each q
//    row's scores lane-local (C col = lane&31 = q row) — the online
//    softmax is pure in-register VALU + ONE shfl_xor(32) per reduction
//    (v2's cross-lane softmax + P-LDS round trip measured 14 VALU insts
//    per MFMA; this removes both).
//  * P -> PV A-fragments in-register: pack bf16 pairs + permlane32_swap
//    half-exchange (guide T12); PV = mfma_32x32x16(A=P, B=V^T read).
//  * C-fragment row map (32x32): row = (r&3) + 8*(r>>2) + 4*(lane>>5),
//    col = lane&31.
//
// Capability parity: the paged-attention prefill the reference delegates
// to vLLM.

#include "common.h"

#include <type_traits>

namespace fi {

typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(16))) float floatx16;

constexpr int kWaves = 8;
constexpr int kQPerWave = 32;                 // q rows per wave (1 MFMA block)
constexpr int kQBlock = kWaves * kQPerWave;   // 256 q rows per workgroup
constexpr int kKVTile = 64;                   // kv tokens per LDS tile
constexpr float kPNegInf = -1e30f;
// defer-max threshold (guide T13): skip the O/l rescale while the tile max
// grows by <= this much — P is then bounded by e^8, which f32 accumulation
// absorbs (max-abs error ~3x vs THR=0 per the guide's measurement; the
// spiked-scores test forces the rescale branch). Scores are in BASE-2
// units (q pre-scaled by log2e), so the e^8 bound is 8*log2e here.
constexpr float kDeferThr = 11.5416f;

// byte-address XOR swizzles, bijective within a ROWB-byte row (guide T2/G4)
template <int ROWB>
FI_DEV int swz(int row, int byte_in_row) {
  constexpr int kMask = (ROWB / 16 > 16 ? 16 : ROWB / 16) - 1;
  return row * ROWB + (byte_in_row ^ ((row & kMask) << 4));
}

// C-fragment row for reg r in a 32x32 MFMA (hi1 = lane>>5)
FI_DEV constexpr int crow(int r, int hi1) {
  return (r & 3) + 8 * (r >> 2) + 4 * hi1;
}

FI_DEV u32 pack_bf16x2(float lo, float hi) {
  return static_cast<u32>(f32_to_bf16(lo)) |
         (static_cast<u32>(f32_to_bf16(hi)) << 16);
}

// FP8 (PAGED only): the cache holds OCP e4m3 at scale 1.0. K cannot use
// global_load_lds (the LDS tile must stay bf16 for the MFMA fragments), so
// both K and V go through the V-style register-stage pipeline with a
// v_cvt_pk_f32_fp8 dequant between load and ds_write — half the HBM bytes
// per tile for ~8 extra VALU ops per 16 elements.
// WPS = target waves per SIMD (__launch_bounds__ 2nd arg, guide §1).
// Measured: WPS=3 spills ~55 slots and LOSES 40% — the kernel is
// register-bound at 2 waves/SIMD; keep 2.
// NW = waves per workgroup (q rows = NW*32). 8 = round-1 shape; 4 halves
// the causal-raggedness idle (a wave's tiles stop at its own diagonal
// but it still barriers with later waves) at the cost of re-reading K/V
// tiles from L2 by 2x more workgroups — A/B'd on hardware (FI_PF_NW).
template <int D, bool PAGED, bool FP8, int WPS = 2, int NW = kWaves>
__global__ __launch_bounds__(NW * kWaveSize, WPS) void prefill_attn_kernel(
    u16* __restrict__ out,        // [T, Hq, D]
    const u16* __restrict__ q,    // [T] rows, stride q_stride
    const void* __restrict__ k_p, // dense rows OR k_cache when PAGED
    const void* __restrict__ v_p, // dense rows OR v_cache when PAGED
    const int* __restrict__ tile_seq,    // [ntiles] sequence index
    const int* __restrict__ tile_row0,   // [ntiles] first NEW q row
    const int* __restrict__ cu_seqlens,  // [nseqs+1] over NEW tokens
    const int* __restrict__ block_tables,  // [S, max_blocks] (PAGED)
    const int* __restrict__ seq_lens_k,    // [S] total ctx len (PAGED)
    const int max_blocks,
    const int64_t q_stride, const int64_t k_stride, const int64_t v_stride,
    const int num_q_heads, const int num_kv_heads, const float scale) {
  using CT = typename std::conditional<FP8, unsigned char, u16>::type;
  const CT* k = static_cast<const CT*>(k_p);
  const CT* v = static_cast<const CT*>(v_p);
  constexpr int KF = D / 16;           // QK^T 16-deep k-chunks over head dim
  constexpr int CB = D / 32;           // PV output 32-col blocks
  constexpr int kKRowB = D * 2;        // K row bytes (D=128 -> 256)
  // V^T rows at their natural 128 B with the (row&7)<<4 in-row swizzle.
  // Round 1 padded them to 256 B for the conflict-free (row&15)<<4
  // swizzle — but that put the workgroup at 98 KB LDS = ONE block/CU
  // (2 waves/SIMD), and PMC showed the kernel latency-bound there
  // (MfmaUtil 11%, SQ_WAIT dominant). 64 KB -> 2 blocks/CU buys 2x the
  // latency hiding for a measured 2.6% bank-conflict cost on the PV
  // reads.
  constexpr int kVTRowB = 128;

  const int head = blockIdx.y;
  const int kv_head = head / (num_q_heads / num_kv_heads);
  const int tid = threadIdx.x;
  const int wave = tid / kWaveSize;
  const int lane = tid % kWaveSize;
  const int col = lane & 31;           // this lane's q row (within the wave)
  const int hi1 = lane >> 5;

  const int seq = tile_seq[blockIdx.x];
  const int seq_start = cu_seqlens[seq];
  const int seq_len = cu_seqlens[seq + 1] - seq_start;  // NEW q rows
  const int k_len = PAGED ? seq_lens_k[seq] : seq_len;  // total kv rows
  const int ctx_start = k_len - seq_len;
  const int wg_row0 = tile_row0[blockIdx.x];
  const int row0 = wg_row0 + wave * kQPerWave;
  const bool active = row0 < seq_len;
  const int* bt_row = PAGED
      ? block_tables + static_cast<int64_t>(seq) * max_blocks : nullptr;
  auto kv_off = [&](int r, int64_t dense_stride) -> int64_t {
    if (PAGED) {
      const int blk = bt_row[r >> 4];
      return ((static_cast<int64_t>(blk) * num_kv_heads + kv_head) * 16 +
              (r & 15)) * D;
    }
    return (seq_start + r) * dense_stride + static_cast<int64_t>(kv_head) * D;
  };

  // LDS: TRIPLE-buffered K (glds target, prefetch depth 2 so a K tile
  // stays in flight ACROSS the tile barrier — round-2: with 2 buffers
  // every barrier drained the glds queue, the guide's ~20% structural
  // stall) + double-buffered V^T; ONE barrier per tile, raw s_barrier +
  // counted vmcnt for bf16 (guide §5 "pipelining across barriers").
  // ONE __shared__ object (guide §5 ".s-level traps" (a)).
  __shared__ u16 smem[3 * kKVTile * D + 2 * D * (kVTRowB / 2)];
  auto k_lds = [&](int i) -> u16* { return smem + i * (kKVTile * D); };
  u16* vt_lds0 = smem + 3 * kKVTile * D;
  u16* vt_lds1 = vt_lds0 + D * (kVTRowB / 2);

  // ---- Q fragments (the QK^T B operand): lane holds
  // Q[row0+col][f*16 + hi1*8 .. +8] for each 16-deep k-chunk f
  // Q pre-scaled by scale*log2e ONCE: the softmax then runs in base-2
  // (exp2 of the raw MFMA scores) with no per-tile scale multiply and no
  // exp argument multiply — ~64 fewer VALU ops per lane per tile.
  // exp2(s*scale*log2e) == exp(s*scale) exactly as a function; the bf16
  // re-round of q*scale adds ~2^-9 relative score noise (within the
  // kernel's fp32-reference tolerance).
  short8 b_q[KF];
  if (active) {
    const int qr = min(row0 + col, seq_len - 1);
    const u16* qrow = q + (seq_start + qr) * q_stride +
                      static_cast<int64_t>(head) * D;
    const float qs = scale * 1.4426950408889634f;  // log2(e)
#pragma unroll
    for (int f = 0; f < KF; ++f) {
      short8 raw = *reinterpret_cast<const short8*>(qrow + f * 16 + hi1 * 8);
      bf16x8 h = __builtin_bit_cast(bf16x8, raw);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        h.h[j] = f32_to_bf16(bf16_to_f32(h.h[j]) * qs);
      b_q[f] = __builtin_bit_cast(short8, h);
    }
  }

  // online-softmax state: ONE q row per lane (lanes j and j+32 share row j)
  float m_row = kPNegInf, l_row = 0.f;
  floatx16 o_acc[CB];
#pragma unroll
  for (int cb = 0; cb < CB; ++cb)
#pragma unroll
    for (int r = 0; r < 16; ++r) o_acc[cb][r] = 0.f;

  const int wg_q_max = ctx_start + min(wg_row0 + NW * kQPerWave, seq_len) - 1;
  const int num_kv_tiles = wg_q_max / kKVTile + 1;
  const int my_q_max = ctx_start + min(row0 + kQPerWave, seq_len) - 1;

  // ---- staging (same pipeline as v2.1) --------------------------------
  constexpr int kKTileBytes = kKVTile * kKRowB;
  constexpr int kGldsPerWave = kKTileBytes / (NW * kWaveSize * 16);
  auto stage_k_glds = [&](int t, u16* kbuf) {
    const int kv0 = t * kKVTile;
#pragma unroll
    for (int i = 0; i < kGldsPerWave; ++i) {
      const int base_off = wave * (kKTileBytes / NW) + i * (kWaveSize * 16);
      const int X = base_off + lane * 16;
      const int row = X / kKRowB;
      const int sbyte = X % kKRowB;
      constexpr int kMask = (kKRowB / 16 > 16 ? 16 : kKRowB / 16) - 1;
      const int byte = sbyte ^ ((row & kMask) << 4);
      const int src_row = min(kv0 + row, k_len - 1);
      const CT* gsrc = k + kv_off(src_row, k_stride) + byte / sizeof(u16);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)gsrc,
          (__attribute__((address_space(3))) void*)(
              reinterpret_cast<char*>(kbuf) + base_off),
          16, 0, 0);
    }
  };
  constexpr int kChunks = D / 16;
  // each thread owns VITER (c16, token) staging positions so every NW
  // covers the full 64 x D tile (NW=8/D=128: 1; NW=4/D=128: 2)
  constexpr int VITER = ceil_div(kChunks, NW);
  const int v_kv = tid % kKVTile;
  const int v_c0 = tid / kKVTile;  // first 16-dim chunk; step NW per iter
  // 16 fp8 bytes (one uint4) -> two bf16x8
  auto cvt_fp8x16 = [](uint4 raw, bf16x8& lo, bf16x8& hi) {
    const u32 w[4] = {raw.x, raw.y, raw.z, raw.w};
    float f[16];
#pragma unroll
    for (int i = 0; i < 4; ++i) unpack_fp8x4(w[i], &f[i * 4]);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      lo.h[j] = f32_to_bf16(f[j]);
      hi.h[j] = f32_to_bf16(f[8 + j]);
    }
  };
  bf16x8 vreg0[VITER], vreg1[VITER];
  auto vload = [&](int t) {
    const int src = min(t * kKVTile + v_kv, k_len - 1);
    const CT* vrow = v + kv_off(src, v_stride);
#pragma unroll
    for (int it = 0; it < VITER; ++it) {
      const int c16 = v_c0 + it * NW;
      if (c16 >= kChunks) break;
      if (FP8) {
        cvt_fp8x16(*reinterpret_cast<const uint4*>(vrow + c16 * 16),
                   vreg0[it], vreg1[it]);
      } else {
        vreg0[it] = *reinterpret_cast<const bf16x8*>(vrow + c16 * 16);
        vreg1[it] = *reinterpret_cast<const bf16x8*>(
            reinterpret_cast<const u16*>(vrow + c16 * 16) + 8);
      }
    }
  };
  // FP8 K staging: same (row, 16-col chunk) ownership as V, but written
  // UNtransposed into the K tile layout ([64 rows][D] bf16, row swizzle);
  // each converted 16-elem group is one 16-B-aligned ds_write_b128 pair
  bf16x8 kreg0[VITER], kreg1[VITER];
  auto kload_fp8 = [&](int t) {
    const int src = min(t * kKVTile + v_kv, k_len - 1);
    const CT* krow = k + kv_off(src, k_stride);
#pragma unroll
    for (int it = 0; it < VITER; ++it) {
      const int c16 = v_c0 + it * NW;
      if (c16 >= kChunks) break;
      cvt_fp8x16(*reinterpret_cast<const uint4*>(krow + c16 * 16),
                 kreg0[it], kreg1[it]);
    }
  };
  auto kwrite_fp8 = [&](u16* kbuf) {
    char* kbase = reinterpret_cast<char*>(kbuf);
#pragma unroll
    for (int it = 0; it < VITER; ++it) {
      const int c16 = v_c0 + it * NW;
      if (c16 >= kChunks) break;
      *reinterpret_cast<bf16x8*>(
          kbase + swz<kKRowB>(v_kv, (c16 * 16) * 2)) = kreg0[it];
      *reinterpret_cast<bf16x8*>(
          kbase + swz<kKRowB>(v_kv, (c16 * 16 + 8) * 2)) = kreg1[it];
    }
  };
  auto vwrite = [&](u16* vt_buf) {
    char* vbase = reinterpret_cast<char*>(vt_buf);
#pragma unroll
    for (int it = 0; it < VITER; ++it) {
      const int c16 = v_c0 + it * NW;
      if (c16 >= kChunks) break;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        *reinterpret_cast<u16*>(
            vbase + swz<kVTRowB>(c16 * 16 + j, v_kv * 2)) = vreg0[it].h[j];
        *reinterpret_cast<u16*>(
            vbase + swz<kVTRowB>(c16 * 16 + j + 8, v_kv * 2)) = vreg1[it].h[j];
      }
    }
  };

  // hoist the swizzled LDS read offsets (constant per lane) out of the
  // tile loop — the swz() address math was otherwise recomputed per tile
  int k_off[2][KF];
  int v_off[4][CB];
#pragma unroll
  for (int f = 0; f < KF; ++f) {
    k_off[0][f] = swz<kKRowB>(col, f * 32 + hi1 * 16);
    k_off[1][f] = swz<kKRowB>(32 + col, f * 32 + hi1 * 16);
  }
#pragma unroll
  for (int kc = 0; kc < 4; ++kc)
#pragma unroll
    for (int cb = 0; cb < CB; ++cb)
      v_off[kc][cb] = swz<kVTRowB>(cb * 32 + col, kc * 32 + hi1 * 16);

  // outstanding-vmem count a tile barrier leaves in flight: the NEWEST
  // K-tile glds (kGldsPerWave wave ops) + the newest V loads (2 per
  // VITER) — everything older (incl. the K tile consumed next) drains.
  // Staging is UNCONDITIONAL with clamped source rows, so the count is
  // a compile-time constant on every iteration (no ragged tail).
  constexpr int kAhead = kGldsPerWave + 2 * VITER;
  auto tile_barrier = [&] {
    asm volatile("s_waitcnt vmcnt(%0)" ::"n"(kAhead) : "memory");
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
  };

  // prologue: K(0)+K(1) and V(0)+V(1) in flight, vt[0] published
  if (FP8) {
    kload_fp8(0);
    vload(0);
    kwrite_fp8(k_lds(0));
    vwrite(vt_lds0);
    if (num_kv_tiles > 1) {
      kload_fp8(1);
      vload(1);
    }
    __syncthreads();        // publishes K(0) + vt[0]
  } else {
    stage_k_glds(0, k_lds(0));
    vload(0);
    stage_k_glds(1, k_lds(1));
    // drain K(0) (own portion; the barrier below publishes cross-wave)
    asm volatile("s_waitcnt vmcnt(%0)" ::"n"(kAhead) : "memory");
    vwrite(vt_lds0);
    vload(1);
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();   // publishes K(0) + vt[0]
  }

  for (int t = 0; t < num_kv_tiles; ++t) {
    const int kv0 = t * kKVTile;
    u16* kbuf = FP8 ? k_lds(t & 1) : k_lds(t % 3);
    u16* kbuf_next = k_lds((t + 1) & 1);  // fp8 staging target
    const u16* vt_cur = (t & 1) ? vt_lds1 : vt_lds0;
    u16* vt_next = (t & 1) ? vt_lds0 : vt_lds1;

    // depth-2 K prefetch (clamped rows: past-the-end tiles re-read the
    // last row — a few hundred wasted bytes beat a ragged vmcnt count)
    if (!FP8) stage_k_glds(t + 2, k_lds((t + 2) % 3));

    const bool compute = active && kv0 <= my_q_max;
    floatx16 s_acc[2];  // S^T for kv blocks [kv0, +32) and [kv0+32, +64)
    if (compute) {
      const char* kbase = reinterpret_cast<const char*>(kbuf);
#pragma unroll
      for (int b = 0; b < 2; ++b)
#pragma unroll
        for (int r = 0; r < 16; ++r) s_acc[b][r] = 0.f;
      // ---- swapped QK^T: S^T[kv 64 x q 32] ----
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int f = 0; f < KF; ++f) {
        const short8 a_k0 =
            *reinterpret_cast<const short8*>(kbase + k_off[0][f]);
        const short8 a_k1 =
            *reinterpret_cast<const short8*>(kbase + k_off[1][f]);
        s_acc[0] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            a_k0, b_q[f], s_acc[0], 0, 0, 0);
        s_acc[1] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            a_k1, b_q[f], s_acc[1], 0, 0, 0);
      }
      __builtin_amdgcn_s_setprio(0);

      // ---- in-register online softmax (row = col, lane-local) ----
      const int q_pos = ctx_start + row0 + col;     // this lane's q row
      const bool row_valid = row0 + col < seq_len;
      // interior tiles (all kv strictly before every valid q row, no
      // ragged tail, no invalid rows) skip the per-element masking
      const bool need_mask =
          (kv0 + kKVTile - 1 > ctx_start + row0)
          || (kv0 + kKVTile > k_len)
          || (row0 + kQPerWave > seq_len);
      if (need_mask) {
#pragma unroll
        for (int b = 0; b < 2; ++b)
#pragma unroll
          for (int r = 0; r < 16; ++r) {
            const int kv_pos = kv0 + 32 * b + crow(r, hi1);
            if (!row_valid || kv_pos > q_pos || kv_pos >= k_len)
              s_acc[b][r] = kPNegInf;
          }
      }  // interior tiles: scores already scaled (q pre-scale) — no pass
      float rm = kPNegInf;
#pragma unroll
      for (int b = 0; b < 2; ++b)
#pragma unroll
        for (int r = 0; r < 16; ++r) rm = fmaxf(rm, s_acc[b][r]);
      rm = fmaxf(rm, __shfl_xor(rm, 32, 64));
      float alpha;
      if (m_row > kPNegInf && __all(rm - m_row <= kDeferThr)) {
        alpha = 1.f;  // defer-max: keep the old running max, skip rescale
      } else {
        const float m_new = fmaxf(m_row, rm);
        if (m_new <= kPNegInf) {
          alpha = 0.f;  // row has seen no unmasked score yet (o is 0)
        } else {
          alpha = (m_row <= kPNegInf) ? 0.f : __builtin_amdgcn_exp2f(m_row - m_new);
          m_row = m_new;
        }
      }
      float rs = 0.f;
#pragma unroll
      for (int b = 0; b < 2; ++b)
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const float pv = (m_row <= kPNegInf || s_acc[b][r] <= kPNegInf)
                               ? 0.f
                               : __builtin_amdgcn_exp2f(s_acc[b][r] - m_row);
          s_acc[b][r] = pv;  // reuse as P (bounded by e^kDeferThr)
          rs += pv;
        }
      rs += __shfl_xor(rs, 32, 64);
      l_row = l_row * alpha + rs;

      // rescale o: alpha is per-ROW (lane-local); o_acc rows follow the
      // crow map, so gather the right alpha per reg via shfl
      if (__any(alpha != 1.f)) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const float a_r = __shfl(alpha, crow(r, hi1), 64);
#pragma unroll
          for (int cb = 0; cb < CB; ++cb) o_acc[cb][r] *= a_r;
        }
      }
    }

    if (compute) {
      const char* vbase = reinterpret_cast<const char*>(vt_cur);
      // ---- P -> A-fragments in-register (pack pairs + half swap) ----
      // A-frag for kv chunk [32b + 16c, +16): lane needs kv {hi1*8..+8} of
      // that range; own regs hold kv {..}+4*hi1 interleaved, the partner
      // half-wave holds the other 4-run — one permlane32_swap per pair
      // merges them (guide T12).
      short8 a_p[4];
#pragma unroll
      for (int b = 0; b < 2; ++b)
#pragma unroll
        for (int c = 0; c < 2; ++c) {
          const int base = 8 * c;
          const u32 u1 = pack_bf16x2(s_acc[b][base + 0], s_acc[b][base + 1]);
          const u32 u2 = pack_bf16x2(s_acc[b][base + 2], s_acc[b][base + 3]);
          const u32 v1 = pack_bf16x2(s_acc[b][base + 4], s_acc[b][base + 5]);
          const u32 v2 = pack_bf16x2(s_acc[b][base + 6], s_acc[b][base + 7]);
          const auto s1 = __builtin_amdgcn_permlane32_swap(
              static_cast<int>(u1), static_cast<int>(v1), false, false);
          const auto s2 = __builtin_amdgcn_permlane32_swap(
              static_cast<int>(u2), static_cast<int>(v2), false, false);
          typedef __attribute__((ext_vector_type(4))) unsigned int u32x4;
          u32x4 w;
          w.x = static_cast<u32>(s1[0]);
          w.y = static_cast<u32>(s2[0]);
          w.z = static_cast<u32>(s1[1]);
          w.w = static_cast<u32>(s2[1]);
          a_p[2 * b + c] = __builtin_bit_cast(short8, w);
        }

      // ---- PV: O[32 q x D] += P[32 x 64] V[64 x D] ----
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int kc = 0; kc < 4; ++kc) {   // 4 x 16-deep kv chunks
#pragma unroll
        for (int cb = 0; cb < CB; ++cb) {
          const short8 b_v =
              *reinterpret_cast<const short8*>(vbase + v_off[kc][cb]);
          o_acc[cb] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              a_p[kc], b_v, o_acc[cb], 0, 0, 0);
        }
      }
      __builtin_amdgcn_s_setprio(0);
    }
    // stage K/V(t+1) (regs already resident) and prefetch (t+2)
    if (FP8) {
      if (t + 1 < num_kv_tiles) {
        kwrite_fp8(kbuf_next);
        vwrite(vt_next);
        if (t + 2 < num_kv_tiles) {
          kload_fp8(t + 2);
          vload(t + 2);
        }
      }
      // fp8: register-staged K, no glds in flight — plain barrier
      __syncthreads();
    } else {
      vwrite(vt_next);   // unconditional: garbage past the end, never read
      vload(t + 2);      // clamped rows
      // ONE raw barrier: publishes vt[(t+1)&1] (lgkmcnt) and drains
      // K(t+1) (counted vmcnt) while K(t+2)+V(t+2) STAY IN FLIGHT
      // across it — the whole point of the 3rd K buffer
      tile_barrier();
    }
  }

  // ---- epilogue: normalize (l gathered per crow row) and store ----
  if (active) {
    const float inv_own = (l_row > 0.f) ? 1.f / l_row : 0.f;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int qr = row0 + crow(r, hi1);
      if (qr >= seq_len) continue;
      const float inv = __shfl(inv_own, crow(r, hi1), 64);
      u16* orow = out + (static_cast<int64_t>(seq_start + qr) * num_q_heads +
                         head) * D;
#pragma unroll
      for (int cb = 0; cb < CB; ++cb)
        orow[cb * 32 + col] = f32_to_bf16(o_acc[cb][r] * inv);
    }
  }
}

void launch_prefill_attn(u16* out, const u16* q, const void* k, const void* v,
                         const int* tile_seq, const int* tile_row0,
                         const int* cu_seqlens, const int* block_tables,
                         const int* seq_lens_k, int max_blocks, int ntiles,
                         int64_t q_stride, int64_t k_stride, int64_t v_stride,
                         int num_q_heads, int num_kv_heads, int head_dim,
                         float scale, bool fp8, int nw, hipStream_t stream) {
  const bool paged = block_tables != nullptr;
  // nw = waves per workgroup (4 or 8), chosen per batch by the host
  // (ops.prefill_tile_rows: 128-row tiles for long sequences); the tile
  // table must have been built with rows == nw*32
  dim3 grid(ntiles, num_q_heads), block(nw * kWaveSize);
#define FI_PF_LAUNCH(DD, PP, F8)                                              \
  do {                                                                        \
    if (nw == 4)                                                              \
      hipLaunchKernelGGL((prefill_attn_kernel<DD, PP, F8, 2, 4>), grid,       \
                         block, 0, stream, out, q, k, v, tile_seq,            \
                         tile_row0, cu_seqlens, block_tables, seq_lens_k,     \
                         max_blocks, q_stride, k_stride, v_stride,            \
                         num_q_heads, num_kv_heads, scale);                   \
    else                                                                      \
      hipLaunchKernelGGL((prefill_attn_kernel<DD, PP, F8, 2, 8>), grid,       \
                         block, 0, stream, out, q, k, v, tile_seq,            \
                         tile_row0, cu_seqlens, block_tables, seq_lens_k,     \
                         max_blocks, q_stride, k_stride, v_stride,            \
                         num_q_heads, num_kv_heads, scale);                   \
  } while (0)
  if (fp8 && !paged) abort();  // fp8 KV is a cache format; dense is bf16
  if (head_dim == 128) {
    if (paged) {
      if (fp8) FI_PF_LAUNCH(128, true, true);
      else FI_PF_LAUNCH(128, true, false);
    } else {
      FI_PF_LAUNCH(128, false, false);
    }
  } else if (head_dim == 64) {
    if (paged) {
      if (fp8) FI_PF_LAUNCH(64, true, true);
      else FI_PF_LAUNCH(64, true, false);
    } else {
      FI_PF_LAUNCH(64, false, false);
    }
  } else {
    abort();
  }
#undef FI_PF_LAUNCH
}

}  // namespace fi
