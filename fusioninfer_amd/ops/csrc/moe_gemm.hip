// Grouped MFMA GEMM for MoE expert compute (gfx950).
//
// Replaces the round-1 capacity-padded torch.bmm experts (and their
// per-layer `counts.max()` host sync) with a vLLM-style block-aligned
// grouped GEMM: token assignments are sorted by expert and each expert's
// segment is padded to a multiple of BLOCK_M on the DEVICE, so every
// m-tile belongs to exactly one expert and all launch shapes are static
// — the whole MoE decode step becomes hipGraph-capturable.
// (Capability parity: expert serving the reference invokes via vLLM
// images; reference publishes no kernels of its own.)
//
// Design (per /opt/skills/guides/cdna_hip_programming.md):
//  * mfma_f32_16x16x32_bf16 per-wave tiles; expert weights are packed
//    OFFLINE into MFMA B-fragment order [E][K/32][N/16][64][8] so each
//    wave's per-K-step B load is ONE fully-coalesced 16 B/lane
//    global_load_dwordx4 (1 KiB per wave per fragment).
//  * No LDS, no barriers: weights stream once (the decode regime is
//    weight-bandwidth-bound: every active expert's panel is read ~once
//    per step), activations are L2-resident; per the guide's GEMV row,
//    operands streamed once and not shared across waves go straight to
//    VGPRs with a deep unroll.
//  * GEMM1 fuses the SwiGLU epilogue: each wave accumulates the gate
//    AND up columns for its 16-column slice and writes silu(g)*u —
//    the [T, 2I] intermediate never round-trips HBM.
//  * Blocks beyond the real (data-dependent) tile count read a device
//    scalar and exit: grid size is static, work is dynamic.
//
// C-fragment map (16x16): row = (lane>>4)*4 + r, col = lane&15.
// A-fragment: lane holds A[row = lane&15][k = (lane>>4)*8 + e], e<8.
// B-fragment: lane holds B[k = (lane>>4)*8 + e][col = lane&15].

#include <hip/hip_runtime.h>

#include "common.h"

namespace fi {

namespace {

typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(4))) float floatx4;

FI_DEV short8 load_b16x8(const u16* p) {
  return *reinterpret_cast<const short8*>(p);
}

// WM x WN waves (4 total), MITER 16-row fragments per wave, NITER
// 16-col fragments per wave. Block tile: rows BM = 16*WM*MITER, cols
// BN = 16*WN*NITER of the OUTPUT space. NITER amortizes the A-panel
// traffic: every n-tile block re-reads its m-tile's A rows, and at
// 8192-token prefill chunks A is larger than L2 — the v1 profile
// (profiles/r02_moe_grouped_v1.md) showed the block_m=128 GEMMs at 52%
// of GPU time, A-re-read-bound.
// GATE_UP: B is packed over 2N columns (gate at n, up at n + N);
// epilogue writes silu(gate)*up. Otherwise a plain grouped GEMM.
// STAGE_A (BM=128 variants): the block's A tile goes through LDS via
// global_load_lds (double-buffered, T2 source-side swizzle, ONE barrier
// per K-step — the guide's 2-phase grouped-GEMM shape), so the four
// waves share one gather of the 128x32 panel instead of issuing
// per-wave fragment loads against L2/L3.
template <int WM, int WN, int MITER, int NITER, bool GATE_UP,
          bool STAGE_A = false>
__global__ __launch_bounds__(256) void moe_gemm_kernel(
    u16* __restrict__ out,              // [PM, N] bf16
    const u16* __restrict__ a,          // GATE_UP: x [T, K]; else act [PM, K]
    const u16* __restrict__ b,          // [E][K/32][NB/16][64][8] bf16
    const int* __restrict__ sorted_ids, // [PM] token row per padded slot
    const int* __restrict__ expert_ids, // [PM/BM] local expert per m-tile
    const int* __restrict__ n_valid,    // device scalar: real m-tile count
    const int K, const int N) {
  constexpr int BM = 16 * WM * MITER;
  static_assert(!STAGE_A || BM == 128, "STAGE_A is tuned for BM=128");
  const int mtile = blockIdx.x;
  if (mtile >= *n_valid) return;

  const int lane = threadIdx.x % kWaveSize;
  const int wave = threadIdx.x / kWaveSize;
  const int wm = wave / WN;
  const int wn = wave % WN;
  // wave's first 16-col fragment index in [0, N/16)
  const int nt0 = (blockIdx.y * WN + wn) * NITER;
  const int e = expert_ids[mtile];
  const int row0 = mtile * BM + wm * (16 * MITER);

  const int NB16 = (GATE_UP ? 2 * N : N) / 16;  // packed B n-fragments
  const u16* b_e =
      b + static_cast<int64_t>(e) * (K / 32) * NB16 * (64 * 8);
  // per-K-step fragment pointers (advance by NB16*512 elements per kt)
  const u16* bg_p[NITER];
  const u16* bu_p[NITER];
#pragma unroll
  for (int ni = 0; ni < NITER; ++ni) {
    bg_p[ni] = b_e + (static_cast<int64_t>(nt0 + ni) * 64 + lane) * 8;
    bu_p[ni] = GATE_UP
        ? b_e + ((static_cast<int64_t>(nt0 + ni) + N / 16) * 64 + lane) * 8
        : nullptr;
  }
  const int64_t b_step = static_cast<int64_t>(NB16) * 64 * 8;

  // A row pointers: gather through sorted_ids for GEMM1 (padding slots
  // carry a valid dummy row — garbage rows are never read by combine)
  const u16* a_p[MITER];
#pragma unroll
  for (int mi = 0; mi < MITER; ++mi) {
    const int slot = row0 + mi * 16 + (lane & 15);
    const int row = GATE_UP ? sorted_ids[slot] : slot;
    a_p[mi] = a + static_cast<int64_t>(row) * K + (lane >> 4) * 8;
  }

  // ---- STAGE_A machinery (BM=128): [128 rows][64 B] tile, rows
  // swizzled byte^=(row&3)<<4 (T2; glds dest is lane-linear so the
  // swizzle rides on the SOURCE address, rule 21)
  constexpr int kARowB = 64;                    // 32 bf16 per row
  constexpr int kATileB = 128 * kARowB;         // 8 KiB
  __shared__ u16 a_lds[STAGE_A ? 2 * kATileB / 2 : 1];
  const int tid = threadIdx.x;
  // this thread's two 16-B staging pieces per K-step
  int stage_src_off[2];   // element offset within the source row
  int64_t stage_row_off[2];
  int stage_dst[2];
  if (STAGE_A) {
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      const int X = (i * 256 + tid) * 16;       // linear LDS byte
      const int srow = X / kARowB;
      const int sbyte = (X % kARowB) ^ ((srow & 3) << 4);
      const int gr = GATE_UP ? sorted_ids[mtile * BM + srow]
                             : mtile * BM + srow;
      stage_row_off[i] = static_cast<int64_t>(gr) * K;
      stage_src_off[i] = sbyte / 2;
      stage_dst[i] = X / 2;
    }
  }
  auto stage_a = [&](int kt, int buf) {
    if (!STAGE_A) return;
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      const u16* src = a + stage_row_off[i] + kt * 32 + stage_src_off[i];
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)src,
          (__attribute__((address_space(3))) void*)(
              a_lds + buf * (kATileB / 2) + stage_dst[i]),
          16, 0, 0);
    }
  };
  // LDS read offsets for this wave's MITER A fragments (elements)
  int a_lds_off[MITER];
  if (STAGE_A) {
#pragma unroll
    for (int mi = 0; mi < MITER; ++mi) {
      const int srow = wm * (16 * MITER) + mi * 16 + (lane & 15);
      const int byte = ((lane >> 4) * 16) ^ ((srow & 3) << 4);
      a_lds_off[mi] = (srow * kARowB + byte) / 2;
    }
    stage_a(0, 0);
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
  }

  floatx4 acc_g[MITER][NITER];
  floatx4 acc_u[GATE_UP ? MITER : 1][NITER];
#pragma unroll
  for (int mi = 0; mi < MITER; ++mi)
#pragma unroll
    for (int ni = 0; ni < NITER; ++ni) {
      acc_g[mi][ni] = {0.f, 0.f, 0.f, 0.f};
      if (GATE_UP) acc_u[mi][ni] = {0.f, 0.f, 0.f, 0.f};
    }

  const int ksteps = K / 32;
  if (STAGE_A) {
    for (int kt = 0; kt < ksteps; ++kt) {
      if (kt + 1 < ksteps) stage_a(kt + 1, (kt + 1) & 1);
      short8 av[MITER];
      const u16* abuf = a_lds + (kt & 1) * (kATileB / 2);
#pragma unroll
      for (int mi = 0; mi < MITER; ++mi)
        av[mi] = load_b16x8(abuf + a_lds_off[mi]);
#pragma unroll
      for (int ni = 0; ni < NITER; ++ni) {
        const short8 bg = load_b16x8(bg_p[ni] + kt * b_step);
        short8 bu;
        if (GATE_UP) bu = load_b16x8(bu_p[ni] + kt * b_step);
#pragma unroll
        for (int mi = 0; mi < MITER; ++mi) {
          acc_g[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              av[mi], bg, acc_g[mi][ni], 0, 0, 0);
          if (GATE_UP)
            acc_u[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                av[mi], bu, acc_u[mi][ni], 0, 0, 0);
        }
      }
      // 2-phase barrier: drains this step's prefetch glds (next tile
      // becomes readable) and closes the LDS reads of the current one
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __syncthreads();
    }
  } else {
    // register double-buffered software pipeline: the NEXT K-step's
    // A/B fragments stay in flight while this step's MFMAs issue. The
    // decode-regime variants (BM=16, no LDS) were LATENCY-bound at
    // ~50% of the weight-BW roofline with the single-buffer loop (one
    // load batch outstanding per iteration). Manual unroll-by-2 keeps
    // the buffer index compile-time (runtime-indexed register arrays
    // spill to scratch — guide rule 20).
    constexpr int NU = GATE_UP ? NITER : 1;
    short8 av0[MITER], av1[MITER];
    short8 bg0[NITER], bg1[NITER];
    short8 bu0[NU], bu1[NU];
    auto issue = [&](short8 (&avd)[MITER], short8 (&bgd)[NITER],
                     short8 (&bud)[NU], int kt) {
#pragma unroll
      for (int mi = 0; mi < MITER; ++mi)
        avd[mi] = load_b16x8(a_p[mi] + kt * 32);
#pragma unroll
      for (int ni = 0; ni < NITER; ++ni) {
        bgd[ni] = load_b16x8(bg_p[ni] + kt * b_step);
        if (GATE_UP) bud[ni] = load_b16x8(bu_p[ni] + kt * b_step);
      }
    };
    auto consume = [&](short8 (&avv)[MITER], short8 (&bgv)[NITER],
                       short8 (&buv)[NU]) {
#pragma unroll
      for (int ni = 0; ni < NITER; ++ni)
#pragma unroll
        for (int mi = 0; mi < MITER; ++mi) {
          acc_g[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              avv[mi], bgv[ni], acc_g[mi][ni], 0, 0, 0);
          if (GATE_UP)
            acc_u[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                avv[mi], buv[ni % NU], acc_u[mi][ni], 0, 0, 0);
        }
    };
    issue(av0, bg0, bu0, 0);
    for (int kt = 0; kt < ksteps; kt += 2) {
      if (kt + 1 < ksteps) issue(av1, bg1, bu1, kt + 1);
      consume(av0, bg0, bu0);
      if (kt + 1 < ksteps) {
        if (kt + 2 < ksteps) issue(av0, bg0, bu0, kt + 2);
        consume(av1, bg1, bu1);
      }
    }
  }

#pragma unroll
  for (int ni = 0; ni < NITER; ++ni) {
    const int col = (nt0 + ni) * 16 + (lane & 15);
#pragma unroll
    for (int mi = 0; mi < MITER; ++mi) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int orow = row0 + mi * 16 + (lane >> 4) * 4 + r;
        float v;
        if (GATE_UP) {
          const float g = acc_g[mi][ni][r];
          const float u = acc_u[mi][ni][r];
          v = (g / (1.f + __expf(-g))) * u;
        } else {
          v = acc_g[mi][ni][r];
        }
        out[static_cast<int64_t>(orow) * N + col] = f32_to_bf16(v);
      }
    }
  }
}

// fp8 grouped GEMM: A (e4m3, per-row scales from the fused-norm / row
// quant) x B (e4m3, per-expert per-out-channel scales) on
// mfma_f32_16x16x32_fp8_fp8 — HALF the weight bytes of bf16 in the
// weight-bandwidth-bound decode regime, and no per-expert host loops:
// the fp8 MoE path becomes hipGraph-capturable (round 1 ran per-expert
// torch._scaled_mm with .nonzero() syncs). Same block-aligned layout
// and epilogue structure as the bf16 kernel; dequant happens in the
// epilogue: v = acc * a_scale[row] * b_scale[col].
typedef long long i64;

template <int WM, int WN, int MITER, int NITER, bool GATE_UP,
          bool STAGE_A = false>
__global__ __launch_bounds__(256) void moe_gemm_fp8_kernel(
    u16* __restrict__ out,               // [PM, N] bf16
    const unsigned char* __restrict__ a, // e4m3 [T, K] or act8 [PM, K]
    const float* __restrict__ a_scales,  // [T] / [PM] per-row
    const unsigned char* __restrict__ b, // [E][K/32][NB/16][64][8] e4m3
    const float* __restrict__ b_scales,  // [E][NB] per out channel
    const int* __restrict__ sorted_ids,
    const int* __restrict__ expert_ids,
    const int* __restrict__ n_valid,
    const int K, const int N) {
  constexpr int BM = 16 * WM * MITER;
  static_assert(!STAGE_A || BM == 128, "STAGE_A is tuned for BM=128");
  const int mtile = blockIdx.x;
  if (mtile >= *n_valid) return;

  const int lane = threadIdx.x % kWaveSize;
  const int wave = threadIdx.x / kWaveSize;
  const int wm = wave / WN;
  const int wn = wave % WN;
  const int nt0 = (blockIdx.y * WN + wn) * NITER;
  const int e = expert_ids[mtile];
  const int row0 = mtile * BM + wm * (16 * MITER);

  const int NB16 = (GATE_UP ? 2 * N : N) / 16;
  const unsigned char* b_e =
      b + static_cast<int64_t>(e) * (K / 32) * NB16 * (64 * 8);
  const float* bs_e = b_scales + static_cast<int64_t>(e) * NB16 * 16;
  const unsigned char* bg_p[NITER];
  const unsigned char* bu_p[NITER];
#pragma unroll
  for (int ni = 0; ni < NITER; ++ni) {
    bg_p[ni] = b_e + (static_cast<int64_t>(nt0 + ni) * 64 + lane) * 8;
    bu_p[ni] = GATE_UP
        ? b_e + ((static_cast<int64_t>(nt0 + ni) + N / 16) * 64 + lane) * 8
        : nullptr;
  }
  const int64_t b_step = static_cast<int64_t>(NB16) * 64 * 8;

  const unsigned char* a_p[MITER];
  int a_row[MITER];
#pragma unroll
  for (int mi = 0; mi < MITER; ++mi) {
    const int slot = row0 + mi * 16 + (lane & 15);
    a_row[mi] = GATE_UP ? sorted_ids[slot] : slot;
    a_p[mi] = a + static_cast<int64_t>(a_row[mi]) * K + (lane >> 4) * 8;
  }

  // STAGE_A (BM=128): [128 rows][32 B] e4m3 tile through LDS via glds
  // (2-phase; swizzle byte^=(row&1)<<4 on the SOURCE — rule 21)
  constexpr int kARowB = 32;                    // 32 e4m3 per row
  constexpr int kATileB = 128 * kARowB;         // 4 KiB
  __shared__ unsigned char a8_lds[STAGE_A ? 2 * kATileB : 1];
  const int tid = threadIdx.x;
  int64_t stage_row_off = 0;
  int stage_src_off = 0, stage_dst = 0;
  if (STAGE_A) {
    const int X = tid * 16;                     // one 16-B piece per thread
    const int srow = X / kARowB;
    const int sbyte = (X % kARowB) ^ ((srow & 1) << 4);
    const int gr = GATE_UP ? sorted_ids[mtile * BM + srow]
                           : mtile * BM + srow;
    stage_row_off = static_cast<int64_t>(gr) * K;
    stage_src_off = sbyte;
    stage_dst = X;
  }
  auto stage_a = [&](int kt, int buf) {
    if (!STAGE_A) return;
    const unsigned char* src = a + stage_row_off + kt * 32 + stage_src_off;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)src,
        (__attribute__((address_space(3))) void*)(
            a8_lds + buf * kATileB + stage_dst),
        16, 0, 0);
  };
  int a_lds_off[MITER];
  if (STAGE_A) {
#pragma unroll
    for (int mi = 0; mi < MITER; ++mi) {
      const int srow = wm * (16 * MITER) + mi * 16 + (lane & 15);
      const int byte = ((lane >> 4) * 8) ^ ((srow & 1) << 4);
      a_lds_off[mi] = srow * kARowB + byte;
    }
    stage_a(0, 0);
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
  }

  floatx4 acc_g[MITER][NITER];
  floatx4 acc_u[GATE_UP ? MITER : 1][NITER];
#pragma unroll
  for (int mi = 0; mi < MITER; ++mi)
#pragma unroll
    for (int ni = 0; ni < NITER; ++ni) {
      acc_g[mi][ni] = {0.f, 0.f, 0.f, 0.f};
      if (GATE_UP) acc_u[mi][ni] = {0.f, 0.f, 0.f, 0.f};
    }

  const int ksteps = K / 32;
  if (STAGE_A) {
    for (int kt = 0; kt < ksteps; ++kt) {
      if (kt + 1 < ksteps) stage_a(kt + 1, (kt + 1) & 1);
      const unsigned char* abuf = a8_lds + (kt & 1) * kATileB;
      i64 av[MITER];
#pragma unroll
      for (int mi = 0; mi < MITER; ++mi)
        av[mi] = *reinterpret_cast<const i64*>(abuf + a_lds_off[mi]);
#pragma unroll
      for (int ni = 0; ni < NITER; ++ni) {
        const i64 bg = *reinterpret_cast<const i64*>(bg_p[ni] + kt * b_step);
        i64 bu = 0;
        if (GATE_UP)
          bu = *reinterpret_cast<const i64*>(bu_p[ni] + kt * b_step);
#pragma unroll
        for (int mi = 0; mi < MITER; ++mi) {
          acc_g[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(
              av[mi], bg, acc_g[mi][ni], 0, 0, 0);
          if (GATE_UP)
            acc_u[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(
                av[mi], bu, acc_u[mi][ni], 0, 0, 0);
        }
      }
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __syncthreads();
    }
  } else {
    // register double-buffered pipeline (see the bf16 kernel: the
    // BM=16 decode variants are latency-bound without it; unroll-by-2
    // keeps buffer indices compile-time — rule 20)
    constexpr int NU = GATE_UP ? NITER : 1;
    i64 av0[MITER], av1[MITER], bg0[NITER], bg1[NITER], bu0[NU], bu1[NU];
    auto issue = [&](i64 (&avd)[MITER], i64 (&bgd)[NITER],
                     i64 (&bud)[NU], int kt) {
#pragma unroll
      for (int mi = 0; mi < MITER; ++mi)
        avd[mi] = *reinterpret_cast<const i64*>(a_p[mi] + kt * 32);
#pragma unroll
      for (int ni = 0; ni < NITER; ++ni) {
        bgd[ni] = *reinterpret_cast<const i64*>(bg_p[ni] + kt * b_step);
        if (GATE_UP)
          bud[ni] = *reinterpret_cast<const i64*>(bu_p[ni] + kt * b_step);
      }
    };
    auto consume = [&](i64 (&avv)[MITER], i64 (&bgv)[NITER],
                       i64 (&buv)[NU]) {
#pragma unroll
      for (int ni = 0; ni < NITER; ++ni)
#pragma unroll
        for (int mi = 0; mi < MITER; ++mi) {
          acc_g[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(
              avv[mi], bgv[ni], acc_g[mi][ni], 0, 0, 0);
          if (GATE_UP)
            acc_u[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(
                avv[mi], buv[ni % NU], acc_u[mi][ni], 0, 0, 0);
        }
    };
    issue(av0, bg0, bu0, 0);
    for (int kt = 0; kt < ksteps; kt += 2) {
      if (kt + 1 < ksteps) issue(av1, bg1, bu1, kt + 1);
      consume(av0, bg0, bu0);
      if (kt + 1 < ksteps) {
        if (kt + 2 < ksteps) issue(av0, bg0, bu0, kt + 2);
        consume(av1, bg1, bu1);
      }
    }
  }

#pragma unroll
  for (int ni = 0; ni < NITER; ++ni) {
    const int col = (nt0 + ni) * 16 + (lane & 15);
    const float bs_g = bs_e[col];
    const float bs_u = GATE_UP ? bs_e[N + col] : 0.f;
#pragma unroll
    for (int mi = 0; mi < MITER; ++mi) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int oslot = row0 + mi * 16 + (lane >> 4) * 4 + r;
        const int arow = GATE_UP ? sorted_ids[oslot] : oslot;
        const float as = a_scales[arow];
        float v;
        if (GATE_UP) {
          const float g = acc_g[mi][ni][r] * as * bs_g;
          const float u = acc_u[mi][ni][r] * as * bs_u;
          v = (g / (1.f + __expf(-g))) * u;
        } else {
          v = acc_g[mi][ni][r] * as * bs_g;
        }
        out[static_cast<int64_t>(oslot) * N + col] = f32_to_bf16(v);
      }
    }
  }
}

// out[t, :] = sum_k w[t,k] * y[pos[t,k], :]   (pos < 0 -> non-local expert
// or padding: skipped). Deterministic — no atomics, so token-exact TP/PD
// tests stay reproducible.
__global__ void moe_combine_kernel(u16* __restrict__ out,
                                   const u16* __restrict__ y,
                                   const int* __restrict__ pos,
                                   const float* __restrict__ w,
                                   const int topk, const int H) {
  const int t = blockIdx.y;
  const int c0 = (blockIdx.x * blockDim.x + threadIdx.x) * 8;
  if (c0 + 8 > H) return;
  float acc[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
  for (int k = 0; k < topk; ++k) {
    const int p = pos[t * topk + k];
    if (p < 0) continue;
    const float wk = w[t * topk + k];
    const bf16x8 v = *reinterpret_cast<const bf16x8*>(
        y + static_cast<int64_t>(p) * H + c0);
#pragma unroll
    for (int i = 0; i < 8; ++i) acc[i] += wk * bf16_to_f32(v.h[i]);
  }
  bf16x8 o;
#pragma unroll
  for (int i = 0; i < 8; ++i) o.h[i] = f32_to_bf16(acc[i]);
  *reinterpret_cast<bf16x8*>(out + static_cast<int64_t>(t) * H + c0) = o;
}

}  // namespace

// block_m 16: decode regime (few rows per expert; WM=1,WN=4,MITER=1 —
// weight-streaming-bound, A is tiny/L2-resident).
// block_m 128: prefill regime (WM=2,WN=2,MITER=4) with NITER 2 (gate_up,
// BN=64) / 4 (down, BN=128) so the block's A-panel read amortizes over
// 4x the output columns of v1.
void launch_moe_gemm(u16* out, const u16* a, const u16* b,
                     const int* sorted_ids, const int* expert_ids,
                     const int* n_valid, int max_mtiles, int K, int N,
                     int block_m, bool gate_up, hipStream_t stream) {
  if (block_m == 16) {
    const dim3 grid(max_mtiles, N / 64);
    if (gate_up)
      hipLaunchKernelGGL((moe_gemm_kernel<1, 4, 1, 1, true>), grid, dim3(256),
                         0, stream, out, a, b, sorted_ids, expert_ids,
                         n_valid, K, N);
    else
      hipLaunchKernelGGL((moe_gemm_kernel<1, 4, 1, 1, false>), grid,
                         dim3(256), 0, stream, out, a, b, sorted_ids,
                         expert_ids, n_valid, K, N);
  } else {  // block_m == 128
    // FI_MOE_STAGE_A=0 falls back to the register-gather form (A/B knob)
    static const bool stage_a = [] {
      const char* e = getenv("FI_MOE_STAGE_A");
      return !(e && e[0] == '0');
    }();
    if (gate_up) {
      const dim3 grid(max_mtiles, N / 64);
      if (stage_a)
        hipLaunchKernelGGL((moe_gemm_kernel<2, 2, 4, 2, true, true>), grid,
                           dim3(256), 0, stream, out, a, b, sorted_ids,
                           expert_ids, n_valid, K, N);
      else
        hipLaunchKernelGGL((moe_gemm_kernel<2, 2, 4, 2, true, false>), grid,
                           dim3(256), 0, stream, out, a, b, sorted_ids,
                           expert_ids, n_valid, K, N);
    } else {
      const dim3 grid(max_mtiles, N / 128);
      if (stage_a)
        hipLaunchKernelGGL((moe_gemm_kernel<2, 2, 4, 4, false, true>), grid,
                           dim3(256), 0, stream, out, a, b, sorted_ids,
                           expert_ids, n_valid, K, N);
      else
        hipLaunchKernelGGL((moe_gemm_kernel<2, 2, 4, 4, false, false>), grid,
                           dim3(256), 0, stream, out, a, b, sorted_ids,
                           expert_ids, n_valid, K, N);
    }
  }
}

void launch_moe_gemm_fp8(u16* out, const unsigned char* a,
                         const float* a_scales, const unsigned char* b,
                         const float* b_scales, const int* sorted_ids,
                         const int* expert_ids, const int* n_valid,
                         int max_mtiles, int K, int N, int block_m,
                         bool gate_up, hipStream_t stream) {
  if (block_m == 16) {
    const dim3 grid(max_mtiles, N / 64);
    if (gate_up)
      hipLaunchKernelGGL((moe_gemm_fp8_kernel<1, 4, 1, 1, true>), grid,
                         dim3(256), 0, stream, out, a, a_scales, b, b_scales,
                         sorted_ids, expert_ids, n_valid, K, N);
    else
      hipLaunchKernelGGL((moe_gemm_fp8_kernel<1, 4, 1, 1, false>), grid,
                         dim3(256), 0, stream, out, a, a_scales, b, b_scales,
                         sorted_ids, expert_ids, n_valid, K, N);
  } else {
    static const bool stage_a8 = [] {
      const char* e = getenv("FI_MOE_STAGE_A");
      return !(e && e[0] == '0');
    }();
    if (gate_up) {
      const dim3 grid(max_mtiles, N / 64);
      if (stage_a8)
        hipLaunchKernelGGL((moe_gemm_fp8_kernel<2, 2, 4, 2, true, true>),
                           grid, dim3(256), 0, stream, out, a, a_scales, b,
                           b_scales, sorted_ids, expert_ids, n_valid, K, N);
      else
        hipLaunchKernelGGL((moe_gemm_fp8_kernel<2, 2, 4, 2, true, false>),
                           grid, dim3(256), 0, stream, out, a, a_scales, b,
                           b_scales, sorted_ids, expert_ids, n_valid, K, N);
    } else {
      const dim3 grid(max_mtiles, N / 128);
      if (stage_a8)
        hipLaunchKernelGGL((moe_gemm_fp8_kernel<2, 2, 4, 4, false, true>),
                           grid, dim3(256), 0, stream, out, a, a_scales, b,
                           b_scales, sorted_ids, expert_ids, n_valid, K, N);
      else
        hipLaunchKernelGGL((moe_gemm_fp8_kernel<2, 2, 4, 4, false, false>),
                           grid, dim3(256), 0, stream, out, a, a_scales, b,
                           b_scales, sorted_ids, expert_ids, n_valid, K, N);
    }
  }
}

void launch_moe_combine(u16* out, const u16* y, const int* pos,
                        const float* w, int tokens, int topk, int H,
                        hipStream_t stream) {
  const int threads = 64;
  const dim3 grid(ceil_div(H / 8, threads), tokens);
  hipLaunchKernelGGL(moe_combine_kernel, grid, dim3(threads), 0, stream, out,
                     y, pos, w, topk, H);
}

// Device-side MoE block alignment in ONE kernel (replaces the ~12-op
// torch composition in MoEMLP._moe_align: scatter_add histogram,
// argsort, cumsums, searchsorted, scatters — measured ~6% of 30B MoE
// GPU time plus a dozen eager launches per layer during prefill).
// Single workgroup: ≤16K assignments and ≤128 local experts make LDS
// histograms + __syncthreads phases trivial. Within-expert slot order
// is atomicAdd arrival order (NOT the stable argsort order) — the
// grouped GEMM computes each row's K-loop identically wherever it sits
// in the expert's segment and the combine gathers by pos, so outputs
// are bit-identical regardless of slot permutation. vLLM parity:
// moe_align_block_size (csrc/moe/moe_align_sum_kernels.cu).
namespace {
__global__ void moe_align_kernel(
    const int* __restrict__ topi,  // [n] = T*topk flattened GLOBAL ids
    int* __restrict__ sorted_ids,  // [PM]   (zeroed here; pad rows -> 0)
    int* __restrict__ expert_ids,  // [PM / block_m]
    int* __restrict__ n_valid,     // [1] real m-tile count
    int* __restrict__ pos,         // [n]  padded slot (-1 = non-local)
    const int n, const int topk, const int e_start, const int e_end,
    const int block_m, const int PM) {
  const int E = e_end - e_start;  // <= 128 by LDS sizing below
  __shared__ int cnt[129], pad0[129], tcum[129], off[129];
  const int tid = threadIdx.x, nt = blockDim.x;
  for (int i = tid; i <= E; i += nt) cnt[i] = 0;
  for (int i = tid; i < PM; i += nt) sorted_ids[i] = 0;
  __syncthreads();
  for (int i = tid; i < n; i += nt) {
    const int e = topi[i];
    const int key = (e >= e_start && e < e_end) ? e - e_start : E;
    atomicAdd(&cnt[key], 1);
  }
  __syncthreads();
  if (tid == 0) {
    int tiles = 0;
    for (int e = 0; e < E; ++e) {
      pad0[e] = tiles * block_m;
      tiles += (cnt[e] + block_m - 1) / block_m;
      tcum[e] = tiles;
    }
    n_valid[0] = tiles;
  }
  __syncthreads();
  // expert of m-tile t = first e with tcum[e] > t (searchsorted
  // right=True), clamped to E-1 for the padded tail
  for (int t = tid; t < PM / block_m; t += nt) {
    int lo = 0, hi = E - 1, ans = E - 1;
    while (lo <= hi) {
      const int mid = (lo + hi) >> 1;
      if (tcum[mid] > t) { ans = mid; hi = mid - 1; }
      else lo = mid + 1;
    }
    expert_ids[t] = ans;
  }
  for (int i = tid; i <= E; i += nt) off[i] = 0;
  __syncthreads();
  for (int i = tid; i < n; i += nt) {
    const int e = topi[i];
    const int key = (e >= e_start && e < e_end) ? e - e_start : E;
    if (key == E) { pos[i] = -1; continue; }
    const int p = pad0[key] + atomicAdd(&off[key], 1);
    pos[i] = p;
    sorted_ids[p] = i / topk;
  }
}
}  // namespace

void launch_moe_align(const int* topi, int* sorted_ids, int* expert_ids,
                      int* n_valid, int* pos, int n, int topk, int e_start,
                      int e_end, int block_m, int PM, hipStream_t stream) {
  hipLaunchKernelGGL(moe_align_kernel, dim3(1), dim3(1024), 0, stream, topi,
                     sorted_ids, expert_ids, n_valid, pos, n, topk, e_start,
                     e_end, block_m, PM);
}

// Fused router tail: softmax + top-k + (optional) renormalize in ONE
// wave-per-token kernel (replaces torch softmax/gatherTopK/bitonic-
// sort/div — ~3.5% of 30B MoE GPU time and 6 launches per layer).
// With renorm (norm_topk_prob), weights = softmax over the top-k
// LOGITS (the global partition cancels); without it, global softmax
// probabilities. Ties pick the smaller expert index.
namespace {
__global__ void moe_router_topk_kernel(
    const float* __restrict__ logits,  // [T, E] fp32
    float* __restrict__ topv,          // [T, k]
    int* __restrict__ topi,            // [T, k]
    const int T, const int E, const int k, const bool renorm) {
  const int t = blockIdx.x * (blockDim.x / kWaveSize)
      + threadIdx.x / kWaveSize;
  if (t >= T) return;
  const int lane = threadIdx.x % kWaveSize;
  const float* row = logits + static_cast<int64_t>(t) * E;
  // two logit slots per lane (E <= 128)
  float l0 = lane < E ? row[lane] : -INFINITY;
  float l1 = lane + 64 < E ? row[lane + 64] : -INFINITY;
  float m = fmaxf(l0, l1);
#pragma unroll
  for (int off = 32; off; off >>= 1) m = fmaxf(m, __shfl_xor(m, off));
  float s = (lane < E ? __expf(l0 - m) : 0.f) +
            (lane + 64 < E ? __expf(l1 - m) : 0.f);
#pragma unroll
  for (int off = 32; off; off >>= 1) s += __shfl_xor(s, off);
  float a0 = l0, a1 = l1;  // remaining candidates
  float ksum = 0.f;
  for (int it = 0; it < k; ++it) {
    float v = a0 >= a1 ? a0 : a1;
    int idx = a0 >= a1 ? lane : lane + 64;
#pragma unroll
    for (int off = 32; off; off >>= 1) {
      const float ov = __shfl_xor(v, off);
      const int oi = __shfl_xor(idx, off);
      if (ov > v || (ov == v && oi < idx)) { v = ov; idx = oi; }
    }
    const float e = __expf(v - m);
    ksum += e;
    if (lane == 0) {
      topi[t * k + it] = idx;
      topv[t * k + it] = renorm ? e : e / s;
    }
    if (idx == lane) a0 = -INFINITY;
    if (idx == lane + 64) a1 = -INFINITY;
  }
  if (renorm && lane == 0) {
    for (int it = 0; it < k; ++it) topv[t * k + it] /= ksum;
  }
}
}  // namespace

void launch_moe_router_topk(const float* logits, float* topv, int* topi,
                            int T, int E, int k, bool renorm,
                            hipStream_t stream) {
  const int waves = 4;  // 4 tokens per 256-thread block
  hipLaunchKernelGGL(moe_router_topk_kernel,
                     dim3(ceil_div(T, waves)), dim3(waves * kWaveSize), 0,
                     stream, logits, topv, topi, T, E, k, renorm);
}

}  // namespace fi
