// Paged-attention DECODE kernels for CDNA4 (gfx950) — v2 (flash-decoding).
//
// One-token attention per sequence against the paged KV cache
// ([num_blocks, Hk, 16, D] bf16). Memory-bound: stream each sequence's K/V
// once at near-HBM rate, reusing every K/V read across the GQA group
// (G q-heads per kv-head).
//
// Grid: (num_seqs, num_kv_heads, num_partitions) — the context is split
// into 512-token partitions (flash-decoding) so small decode batches still
// fill 256 CUs; partial (m, l, acc) go to an fp32 workspace and a second
// kernel merges partitions. num_partitions == 1 writes output directly.
//
// Within a workgroup: 4 waves process interleaved 16-token chunks (1 chunk
// == 1 cache block, so a chunk's K/V are contiguous). Per chunk:
//   phase A (scores): lane = token*4 + i, lane reads K[token][i*32 .. +32]
//     as 4x bf16x8 (64 B); dot with q (fp32 in LDS, read as float4 =
//     ds_read_b128); 2-level shfl_xor reduce; online-softmax state (m, l
//     per G) wave-uniform in registers; probs parked in per-wave LDS.
//   phase B (PV): lane = dim pair, per-token V rows read 256 B coalesced;
//     fp32 accumulator acc[G][2] per lane.
// Cross-wave flash-merge through LDS ends the workgroup.
//
// Capability parity: the paged-attention decode the reference delegates to
// its vLLM containers (SURVEY.md §2.3 "Paged-attention decode kernel").

#include "common.h"

#include <type_traits>

namespace fi {

constexpr int kBlockSz = 16;     // cache block size (tokens)
constexpr int kPartChunks = 32;  // 512 tokens per partition
constexpr float kNegInf = -1e30f;

// kNWaves is a launch-time choice: 8 waves when the grid underfills the
// chip (small decode batches: measured +46% at batch<=8), 4 when it is
// full (4-wave measured ~5% faster there).
// FP8: the cache holds OCP e4m3 at scale 1.0 — half the KV bytes of bf16
// (this kernel is KV-bandwidth-bound); v_cvt_pk_f32_fp8 dequant on load.
// HS (head split): waves are partitioned into HS head-groups of G/HS
// heads each; a wave streams chunks at stride kNWaves/HS for ITS head
// slice only. G=8 runs HS=2 so the per-wave state is the G=4 footprint —
// the monolithic G=8 variant spilled ~500 B/lane (W=8) or sat at
// occupancy 1 (W=4).
typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(4))) float floatx4;

// MFA (bf16 only): phase A runs QK^T on the MATRIX CORES instead of
// per-lane FMAs + 6-7-deep shfl_xor reduction chains (PMC showed the
// old phase A WAIT-dominant on those serial permlanes). One
// mfma_f32_16x16x32_bf16 per 32-dim slice: A = Q fragment with the G2
// head rows REPLICATED over all 16 rows (so every 16-lane col group
// holds a full copy of the scores and the softmax reductions are
// 4-deep within a group, wave-uniform by construction — alpha needs no
// broadcast), B = K fragment (lane reads K[tok=l&15][(l>>4)*8..+8]:
// same bytes as the scalar path, different pattern, block stays
// L2-resident). The K-dim reduction happens INSIDE the MFMA.
template <int D, int G, int kNWaves, bool FP8, int HS = 1,
          bool MFA = false>
__global__ __launch_bounds__(kNWaves * kWaveSize) void paged_attn_decode_kernel(
    u16* __restrict__ out,            // [S, Hq, D] (written when 1 partition)
    float* __restrict__ ml_ws,        // [S, Hq, P, 2] (multi-partition)
    float* __restrict__ acc_ws,       // [S, Hq, P, D]
    const u16* __restrict__ q,        // [S] rows, stride q_stride, Hq*D elems
    const void* __restrict__ k_cache_p,  // [B, Hk, 16, D] bf16 (FP8: e4m3)
    const void* __restrict__ v_cache_p,
    const int* __restrict__ block_tables,  // [S, max_blocks]
    const int* __restrict__ seq_lens,      // [S]
    const int64_t q_stride, const int max_blocks, const int num_kv_heads,
    const float scale) {
  using CT = typename std::conditional<FP8, unsigned char, u16>::type;
  const CT* k_cache = static_cast<const CT*>(k_cache_p);
  const CT* v_cache = static_cast<const CT*>(v_cache_p);
  const int seq = blockIdx.x;
  const int kv_head = blockIdx.y;
  const int part = blockIdx.z;
  const int num_parts = gridDim.z;
  const int tid = threadIdx.x;
  const int wave = tid / kWaveSize;
  const int lane = tid % kWaveSize;
  // head-split decomposition: wave = cwave * HSE + hsplit
  // (HSE clamps HS to G so dead template branches in the launcher's
  // runtime dispatch still compile — e.g. G=1 never runs with HS=2)
  constexpr int HSE = (HS > G) ? 1 : HS;
  constexpr int G2 = G / HSE;         // heads this wave carries
  constexpr int kCWaves = kNWaves / HSE;  // waves streaming chunks together
  const int hsplit = wave % HSE;
  const int cwave = wave / HSE;

  const int ctx = seq_lens[seq];
  const int num_chunks = (ctx + kBlockSz - 1) / kBlockSz;
  const int chunk_lo = part * kPartChunks;
  const int chunk_hi = min(num_chunks, chunk_lo + kPartChunks);
  const int num_heads = num_kv_heads * G;

  // empty partition: publish "nothing" and leave
  if (chunk_lo >= num_chunks) {
    if (tid < G) {
      const int h = kv_head * G + tid;
      float* ml = ml_ws + ((static_cast<int64_t>(seq) * num_heads + h) *
                               num_parts + part) * 2;
      ml[0] = kNegInf;
      ml[1] = 0.f;
    }
    return;
  }

  __shared__ float q_lds[MFA ? 1 : G][MFA ? 1 : D];
  __shared__ float p_lds[kNWaves][kBlockSz][G2];
  __shared__ float merge_m[kNWaves][G2];
  __shared__ float merge_l[kNWaves][G2];
  __shared__ float merge_acc[kNWaves][G2][D];

  // MFA: per-wave replicated-row Q fragments, built once (Q is fixed
  // for the whole kernel). Row i of A carries head i % G2, pre-scaled
  // by scale*log2e and re-rounded to bf16 (same trick as the prefill
  // kernel's pre-scaled Q).
  short8 av_q[(MFA && !FP8) ? D / 32 : 1];
  long long av8_q[(MFA && FP8) ? D / 32 : 1];
  __shared__ float qsc_lds[(MFA && FP8) ? G : 1];
  float qsc_r[G2];  // fp8 MFA: per-head dequant*softmax scalar
  if constexpr (MFA && !FP8) {
    const u16* q_row = q + seq * q_stride +
                       static_cast<int64_t>(kv_head) * G * D;
    const int head = hsplit * G2 + ((lane & 15) % G2);
#pragma unroll
    for (int kk = 0; kk < D / 32; ++kk) {
      const int d0 = kk * 32 + (lane >> 4) * 8;
      u16 tmp[8];
#pragma unroll
      for (int e = 0; e < 8; ++e)
        tmp[e] = f32_to_bf16(bf16_to_f32(q_row[head * D + d0 + e]) *
                             scale * 1.4426950408889634f);
      av_q[kk] = *reinterpret_cast<short8*>(tmp);
    }
  } else if constexpr (MFA && FP8) {
    // Q quantized to e4m3 per HEAD; the dequant scale folds into the
    // post-MFMA per-head scalar together with scale*log2e (the model
    // already folded the cache's k_scale into `scale`)
    const u16* q_row = q + seq * q_stride +
                       static_cast<int64_t>(kv_head) * G * D;
    if (tid < G) {
      float mx = 1e-8f;
      for (int d = 0; d < D; ++d)
        mx = fmaxf(mx, fabsf(bf16_to_f32(q_row[tid * D + d])));
      qsc_lds[tid] = mx / 448.f;
    }
    __syncthreads();
    const int head = hsplit * G2 + ((lane & 15) % G2);
    const float inv_qs = 1.f / qsc_lds[head];
#pragma unroll
    for (int kk = 0; kk < D / 32; ++kk) {
      const int d0 = kk * 32 + (lane >> 4) * 8;
      float f[8];
#pragma unroll
      for (int e = 0; e < 8; ++e)
        f[e] = bf16_to_f32(q_row[head * D + d0 + e]) * inv_qs;
      u32 w[2];
      pack_fp8x8(f, w);
      av8_q[kk] = static_cast<long long>(
          (static_cast<unsigned long long>(w[1]) << 32) | w[0]);
    }
#pragma unroll
    for (int g = 0; g < G2; ++g)
      qsc_r[g] = qsc_lds[hsplit * G2 + g] * scale * 1.4426950408889634f;
  } else {
    // stage q (G heads) into LDS as fp32 (pre-scaled)
    const u16* q_row = q + seq * q_stride +
                       static_cast<int64_t>(kv_head) * G * D;
    for (int e = tid; e < G * D; e += kNWaves * kWaveSize)
      // scale*log2e: scores land in base-2 units so the softmax
      // uses bare v_exp_f32 (exp2) with no argument multiply
      q_lds[e / D][e % D] =
          bf16_to_f32(q_row[e]) * scale * 1.4426950408889634f;
    __syncthreads();
  }

  // online-softmax state, wave-uniform (every lane holds the same copy)
  float m[G2], l[G2], acc[G2][2];
#pragma unroll
  for (int g = 0; g < G2; ++g) {
    m[g] = kNegInf;
    l[g] = 0.f;
    acc[g][0] = acc[g][1] = 0.f;
  }

  const int tok = lane / 4;          // phase-A token within chunk
  const int quad = lane % 4;         // phase-A dim quarter
  constexpr int DPQ = D / 4;         // dims per phase-A lane
  constexpr int KQ4 = DPQ * sizeof(CT) / 16;  // 16-B K loads per lane per chunk

  // software-prefetch: this wave's NEXT chunk's K flies while the current
  // chunk's softmax + PV run (the phases were serialized on K latency)
  auto kv_base_of = [&](int chunk) -> int64_t {
    const int block_id = block_tables[seq * max_blocks + chunk];
    return ((static_cast<int64_t>(block_id) * num_kv_heads + kv_head) *
            kBlockSz) * D;
  };
  uint4 kraw[KQ4];
  auto load_k = [&](int64_t kv_base, uint4* dst) {
    if constexpr (MFA && FP8) {
      // fp8 B-fragment: 8 e4m3 bytes per lane per 32-dim slice
      const CT* k_row = k_cache + kv_base + (lane & 15) * D;
#pragma unroll
      for (int kk = 0; kk < D / 32; ++kk)
        reinterpret_cast<unsigned long long*>(dst)[kk] =
            *reinterpret_cast<const unsigned long long*>(
                k_row + kk * 32 + (lane >> 4) * 8);
    } else if constexpr (MFA) {
      // B-fragment pattern: lane supplies K[tok = l&15][(l>>4)*8 + e]
      const CT* k_row = k_cache + kv_base + (lane & 15) * D;
#pragma unroll
      for (int kk = 0; kk < KQ4; ++kk)
        dst[kk] = *reinterpret_cast<const uint4*>(
            k_row + kk * 32 + (lane >> 4) * 8);
    } else {
      const CT* k_row = k_cache + kv_base + tok * D + quad * DPQ;
#pragma unroll
      for (int j8 = 0; j8 < KQ4; ++j8)
        dst[j8] = reinterpret_cast<const uint4*>(k_row)[j8];
    }
  };
  int64_t kv_base = 0;
  if (chunk_lo + cwave < chunk_hi) {
    kv_base = kv_base_of(chunk_lo + cwave);
    load_k(kv_base, kraw);
  }

  for (int chunk = chunk_lo + cwave; chunk < chunk_hi; chunk += kCWaves) {
    const int token_pos = chunk * kBlockSz + tok;
    const int64_t kv_base_cur = kv_base;

    // ---- phase A: scores for 16 tokens x G2 heads ----
    // streaming convert+FMA: per 16-B K group, convert 8 elements and
    // fold them into all G2 head accumulators immediately — kf[8] live
    // instead of kf[32] (round-2 register diet: the G=4/D=128 variant
    // was 218 VGPR -> 2 waves/SIMD; PMC showed WAIT-dominant)
    float s[G2];
#pragma unroll
    for (int g = 0; g < G2; ++g) s[g] = 0.f;
    if constexpr (MFA) {
      // QK^T on the matrix cores: K-dim reduction inside the MFMA, no
      // cross-lane score reduction needed. c[row=g][col=lane&15].
      floatx4 c = {0.f, 0.f, 0.f, 0.f};
      if constexpr (FP8) {
#pragma unroll
        for (int kk = 0; kk < D / 32; ++kk)
          c = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(
              av8_q[kk],
              static_cast<long long>(
                  reinterpret_cast<const unsigned long long*>(kraw)[kk]),
              c, 0, 0, 0);
#pragma unroll
        for (int g = 0; g < G2; ++g) s[g] = c[g] * qsc_r[g];
      } else {
#pragma unroll
        for (int kk = 0; kk < KQ4; ++kk)
          c = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              av_q[kk], __builtin_bit_cast(short8, kraw[kk]), c, 0, 0, 0);
#pragma unroll
        for (int g = 0; g < G2; ++g) s[g] = c[g];
      }
    } else {
#pragma unroll
      for (int j8 = 0; j8 < KQ4; ++j8) {
        float kf8[16 / sizeof(CT)];
        if (FP8) {
          const u32 w[4] = {kraw[j8].x, kraw[j8].y, kraw[j8].z, kraw[j8].w};
#pragma unroll
          for (int i = 0; i < 4; ++i) unpack_fp8x4(w[i], &kf8[i * 4]);
        } else {
          const bf16x8 kv8 = __builtin_bit_cast(bf16x8, kraw[j8]);
#pragma unroll
          for (int j = 0; j < 8; ++j) kf8[j] = bf16_to_f32(kv8.h[j]);
        }
        constexpr int EPG = 16 / sizeof(CT);  // elems per 16-B group
#pragma unroll
        for (int g = 0; g < G2; ++g) {
#pragma unroll
          for (int j4 = 0; j4 < EPG / 4; ++j4) {
            // float4 -> ds_read_b128 (4x fewer LDS cycles than scalar)
            const float4 qv = *reinterpret_cast<const float4*>(
                &q_lds[hsplit * G2 + g][quad * DPQ + j8 * EPG + j4 * 4]);
            s[g] = fmaf(kf8[j4 * 4 + 0], qv.x, s[g]);
            s[g] = fmaf(kf8[j4 * 4 + 1], qv.y, s[g]);
            s[g] = fmaf(kf8[j4 * 4 + 2], qv.z, s[g]);
            s[g] = fmaf(kf8[j4 * 4 + 3], qv.w, s[g]);
          }
        }
      }
    }
    // preload THIS chunk's V before the softmax so its latency hides
    // under the reductions (it was serialized behind phase A before)
    u32 vball[kBlockSz];
    const CT* v_rows = v_cache + kv_base_cur;
    if (lane < D / 2) {
#pragma unroll
      for (int t = 0; t < kBlockSz; ++t)
        vball[t] = FP8
            ? static_cast<u32>(*reinterpret_cast<const u16*>(
                  v_rows + t * D + 2 * lane))
            : *reinterpret_cast<const u32*>(v_rows + t * D + 2 * lane);
    }
    // issue next chunk's K now; it lands under softmax + PV
    if (chunk + kCWaves < chunk_hi) {
      kv_base = kv_base_of(chunk + kCWaves);
      load_k(kv_base, kraw);
    }
    // fence: stop the scheduler from hoisting phase-B work (and its live
    // ranges) above the softmax — cross-phase overlap doubled VGPRs
    // (G2 is always <= 4: G=8 is head-split)
    __builtin_amdgcn_sched_barrier(0);
    if constexpr (MFA) {
      // this lane's score column is token l&15 (replicated per group)
      const int col_pos = chunk * kBlockSz + (lane & 15);
#pragma unroll
      for (int g = 0; g < G2; ++g)
        if (col_pos >= ctx) s[g] = kNegInf;
    } else {
      // reduce over the 4 dim-quarters (lanes 4t..4t+3)
#pragma unroll
      for (int g = 0; g < G2; ++g) {
        s[g] += __shfl_xor(s[g], 1, 64);
        s[g] += __shfl_xor(s[g], 2, 64);
        if (token_pos >= ctx) s[g] = kNegInf;
      }
    }

    // chunk max over tokens (MFA: 4-deep within the replicated 16-lane
    // col group; scalar path: xor 4..32 spans the 16 token groups)
    float alpha[G2];
#pragma unroll
    for (int g = 0; g < G2; ++g) {
      float cm = s[g];
      if constexpr (MFA) {
#pragma unroll
        for (int off = 1; off < 16; off <<= 1)
          cm = fmaxf(cm, __shfl_xor(cm, off, 64));
      } else {
#pragma unroll
        for (int off = 4; off < 64; off <<= 1)
          cm = fmaxf(cm, __shfl_xor(cm, off, 64));
      }
      const float m_new = fmaxf(m[g], cm);
      alpha[g] = __builtin_amdgcn_exp2f(m[g] - m_new);
      if (m[g] <= kNegInf && m_new <= kNegInf) alpha[g] = 0.f;
      m[g] = m_new;
    }

    // probs + row-sum
#pragma unroll
    for (int g = 0; g < G2; ++g) {
      float p = (s[g] <= kNegInf) ? 0.f
                : __builtin_amdgcn_exp2f(s[g] - m[g]);
      if constexpr (MFA) {
        if (lane < 16) p_lds[wave][lane][g] = p;
        float psum = p;
#pragma unroll
        for (int off = 1; off < 16; off <<= 1)
          psum += __shfl_xor(psum, off, 64);
        l[g] = l[g] * alpha[g] + psum;  // each token counted once
      } else {
        // every token is replicated on 4 lanes -> scale by 1/4
        if (quad == 0) p_lds[wave][tok][g] = p;
        float psum = p;
#pragma unroll
        for (int off = 1; off < 64; off <<= 1)
          psum += __shfl_xor(psum, off, 64);
        l[g] = l[g] * alpha[g] + psum * 0.25f;
      }
    }

    __builtin_amdgcn_sched_barrier(0);
    // ---- phase B: PV accumulate; lane covers dims {2l, 2l+1} ----
    // (vball preloaded before the softmax; p is 0 for padding tokens —
    // phase A masks every invalid position — so the full block is safe)
#pragma unroll
    for (int g = 0; g < G2; ++g) {
      acc[g][0] *= alpha[g];
      acc[g][1] *= alpha[g];
    }
    if (lane < D / 2) {
      for (int t = 0; t < kBlockSz; ++t) {
        float v0, v1;
        if (FP8) {
          float vf[4];
          unpack_fp8x4(vball[t], vf);
          v0 = vf[0];
          v1 = vf[1];
        } else {
          v0 = bf16_to_f32(static_cast<u16>(vball[t] & 0xffff));
          v1 = bf16_to_f32(static_cast<u16>(vball[t] >> 16));
        }
#pragma unroll
        for (int g = 0; g < G2; ++g) {
          const float p = p_lds[wave][t][g];
          acc[g][0] = fmaf(p, v0, acc[g][0]);
          acc[g][1] = fmaf(p, v1, acc[g][1]);
        }
      }
    }
  }

  // ---- cross-wave flash merge (within each head-group) ----
  if (lane < G2) {
    merge_m[wave][lane] = m[lane];
    merge_l[wave][lane] = l[lane];
  }
  if (lane < D / 2) {
#pragma unroll
    for (int g = 0; g < G2; ++g) {
      merge_acc[wave][g][2 * lane] = acc[g][0];
      merge_acc[wave][g][2 * lane + 1] = acc[g][1];
    }
  }
  __syncthreads();

  // head-group hsplit's G2 heads are distributed over its kCWaves waves
#pragma unroll
  for (int g = 0; g < G2; ++g) {
    if ((g % kCWaves) != cwave || lane >= D / 2) continue;
    float gm = kNegInf;
#pragma unroll
    for (int c = 0; c < kCWaves; ++c)
      gm = fmaxf(gm, merge_m[c * HSE + hsplit][g]);
    float L = 0.f, o0 = 0.f, o1 = 0.f;
#pragma unroll
    for (int c = 0; c < kCWaves; ++c) {
      const int w = c * HSE + hsplit;
      const float mw = merge_m[w][g];
      const float f = (mw <= kNegInf) ? 0.f
                      : __builtin_amdgcn_exp2f(mw - gm);
      L += merge_l[w][g] * f;
      o0 = fmaf(merge_acc[w][g][2 * lane], f, o0);
      o1 = fmaf(merge_acc[w][g][2 * lane + 1], f, o1);
    }
    const int h = kv_head * G + hsplit * G2 + g;
    if (num_parts == 1) {
      const float inv = 1.f / L;
      u16* o_row = out + (static_cast<int64_t>(seq) * num_heads + h) * D;
      const u32 obits = static_cast<u32>(f32_to_bf16(o0 * inv)) |
                        (static_cast<u32>(f32_to_bf16(o1 * inv)) << 16);
      *reinterpret_cast<u32*>(o_row + 2 * lane) = obits;
    } else {
      const int64_t slot = (static_cast<int64_t>(seq) * num_heads + h) *
                               gridDim.z + part;
      if (lane == 0) {
        ml_ws[slot * 2] = gm;
        ml_ws[slot * 2 + 1] = L;
      }
      float2* arow = reinterpret_cast<float2*>(acc_ws + slot * D);
      arow[lane] = make_float2(o0, o1);
    }
  }
}

// merge partials: grid (S, Hq), one wave; lane covers dims {2l, 2l+1}
template <int D>
__global__ __launch_bounds__(kWaveSize) void paged_attn_reduce_kernel(
    u16* __restrict__ out,            // [S, Hq, D]
    const float* __restrict__ ml_ws,  // [S, Hq, P, 2]
    const float* __restrict__ acc_ws, // [S, Hq, P, D]
    const int num_parts) {
  const int seq = blockIdx.x;
  const int h = blockIdx.y;
  const int num_heads = gridDim.y;
  const int lane = threadIdx.x;
  if (lane >= D / 2) return;
  const int64_t base = (static_cast<int64_t>(seq) * num_heads + h) * num_parts;

  float gm = kNegInf;
  for (int p = 0; p < num_parts; ++p)
    gm = fmaxf(gm, ml_ws[(base + p) * 2]);
  float L = 0.f, o0 = 0.f, o1 = 0.f;
  for (int p = 0; p < num_parts; ++p) {
    const float mp = ml_ws[(base + p) * 2];
    if (mp <= kNegInf) continue;
    const float f = __builtin_amdgcn_exp2f(mp - gm);
    L += ml_ws[(base + p) * 2 + 1] * f;
    const float2 a =
        reinterpret_cast<const float2*>(acc_ws + (base + p) * D)[lane];
    o0 = fmaf(a.x, f, o0);
    o1 = fmaf(a.y, f, o1);
  }
  const float inv = 1.f / L;
  u16* o_row = out + (static_cast<int64_t>(seq) * num_heads + h) * D;
  const u32 obits = static_cast<u32>(f32_to_bf16(o0 * inv)) |
                    (static_cast<u32>(f32_to_bf16(o1 * inv)) << 16);
  *reinterpret_cast<u32*>(o_row + 2 * lane) = obits;
}

void launch_paged_attn_decode(u16* out, float* ml_ws, float* acc_ws,
                              const u16* q, const void* k_cache,
                              const void* v_cache, const int* block_tables,
                              const int* seq_lens, int num_seqs,
                              int64_t q_stride, int max_blocks,
                              int num_kv_heads, int head_dim, int group,
                              int num_parts, float scale, bool fp8,
                              hipStream_t stream) {
  const bool wide = num_seqs * num_kv_heads * num_parts < 256;
  const int nwaves = wide ? 8 : 4;
  dim3 grid(num_seqs, num_kv_heads, num_parts), block(nwaves * kWaveSize);
  // FI_DECODE_MFMA=0 falls back to the scalar-FMA phase A (both dtypes;
  // the fp8 MFA variant quantizes Q to e4m3 per head)
  static const bool mfa = [] {
    const char* e = getenv("FI_DECODE_MFMA");
    return !(e && e[0] == '0');
  }();
// fp8 MFA dispatch is G=8-only: kbench showed +41-43% there (G2=4
// after head-split) but -21% at G=4 fp8, where the scalar HS=2 path
// with halved KV bytes was already latency-optimal.
#define FI_LAUNCH_1(DD, GG, NW, F8, HSP)                                      \
  if (mfa && (!F8 || GG == 8)) {                                              \
    hipLaunchKernelGGL(                                                       \
        (paged_attn_decode_kernel<DD, GG, NW, F8, HSP, true>), grid,          \
        block, 0, stream, out, ml_ws, acc_ws, q, k_cache, v_cache,            \
        block_tables, seq_lens, q_stride, max_blocks, num_kv_heads, scale);   \
  } else {                                                                    \
    hipLaunchKernelGGL((paged_attn_decode_kernel<DD, GG, NW, F8, HSP>), grid, \
                       block, 0, stream, out, ml_ws, acc_ws, q, k_cache,      \
                       v_cache, block_tables, seq_lens, q_stride, max_blocks, \
                       num_kv_heads, scale);                                  \
  }
// Head-split policy, A/B'd on hardware (2026-09-14 kbench):
// - G=8 always runs HS=2 (monolithic spilled; HS=4 measured a wash at
//   full grids and -12% wide).
// - G=4 runs HS=2 only at FULL grids (4-wave blocks): the G2=2 state
//   compiles to 149 VGPR = 3 waves/SIMD, measured +2.5% bf16 / +8% fp8
//   at b256; at WIDE grids (8-wave, small batches) the halved chunk
//   parallelism lost 12-19%, so wide keeps the monolithic G=4 shape.
// (The streaming-convert phase A + early V preload apply everywhere.)
#define FI_LAUNCH_HS(DD, GG, HSW, HSN)                                        \
  if (fp8) {                                                                  \
    if (wide) { FI_LAUNCH_1(DD, GG, 8, true, HSW); }                          \
    else      { FI_LAUNCH_1(DD, GG, 4, true, HSN); }                          \
  } else {                                                                    \
    if (wide) { FI_LAUNCH_1(DD, GG, 8, false, HSW); }                         \
    else      { FI_LAUNCH_1(DD, GG, 4, false, HSN); }                         \
  }                                                                           \
  if (num_parts > 1) {                                                        \
    dim3 rgrid(num_seqs, num_kv_heads * GG), rblock(kWaveSize);               \
    hipLaunchKernelGGL((paged_attn_reduce_kernel<DD>), rgrid, rblock, 0,      \
                       stream, out, ml_ws, acc_ws, num_parts);                \
  }
#define FI_LAUNCH(DD, GG)                                                     \
  if (GG == 8)      { FI_LAUNCH_HS(DD, GG, 2, 2) }                            \
  else if (GG == 4) { FI_LAUNCH_HS(DD, GG, 1, 2) }                            \
  else              { FI_LAUNCH_HS(DD, GG, 1, 1) }
  if (head_dim == 128) {
    switch (group) {
      case 1: FI_LAUNCH(128, 1); break;
      case 2: FI_LAUNCH(128, 2); break;
      case 4: FI_LAUNCH(128, 4); break;
      case 8: FI_LAUNCH(128, 8); break;
      default: abort();
    }
  } else if (head_dim == 64) {
    switch (group) {
      case 1: FI_LAUNCH(64, 1); break;
      case 2: FI_LAUNCH(64, 2); break;
      case 4: FI_LAUNCH(64, 4); break;
      case 8: FI_LAUNCH(64, 8); break;
      default: abort();
    }
  } else {
    abort();
  }
#undef FI_LAUNCH
#undef FI_LAUNCH_1
}

}  // namespace fi
