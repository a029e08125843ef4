// MFMA decode attention: the GQA query group rides the matrix cores.
//
// Round-2 restructure: decode attention has ZERO K/V reuse inside a workgroup
// (each K row feeds one QK^T fragment, each V^T row feeds one PV fragment),
// so the round-1 LDS staging (and its 32-scalar-store V transpose) was pure
// overhead. The V cache is stored TRANSPOSED ([NB, Hk, D, BS], kv_cache.py),
// which makes both MFMA operand fragments DIRECT 16-byte HBM loads:
//   S^T[KT tok x 16q] = K_tile . Q^T   A-frag: 8 consecutive d of one token
//   O^T[D x 16q]     += V^T . P        A-frag: 8 consecutive tokens of one d
// Q lives in registers; only P and the softmax reductions use LDS, so
// occupancy is bounded by registers, not LDS. K fragments are register
// double-buffered across tiles with NAMED buffers (a runtime-indexed buffer
// array goes to scratch — guide common-mistake #20; the first cut measured
// 272 B/lane of scratch and ran 2x slow). V single-buffers at tile start so
// its latency hides under QK+softmax.
//
// KT (tokens per tile) is a template parameter: 128 halves the per-byte
// barrier/softmax overhead vs 64 at higher register pressure; the launcher
// picks via PARALLAX_DM_KTILE (A/B'd on hardware).
//
// One workgroup per (sequence, kv_head, partition); partitions are fixed-size
// (graph-capture stable) and combine through the shared reduce kernel.
// Sliding window, softcap, sinks, fp8 KV (dequant at load), and the MSA
// SPARSE mode (explicit token-position lists; per-element V gather since
// sparse tokens are not contiguous) ride the same template.

#include "common.h"

#include <stdlib.h>

#define DM_THREADS 256

template <int HEAD_DIM, int BLOCK_SIZE, bool PARTITIONED, bool KV_FP8,
          bool SPARSE = false, int KT = 64>
__global__ __launch_bounds__(DM_THREADS) void paged_decode_mfma_kernel(
    uint16_t* __restrict__ out,        // [B, Hq, D] (final mode)
    float* __restrict__ tmp_acc,       // [B, Hq, P, D] (partitioned)
    float* __restrict__ tmp_ml,        // [B, Hq, P, 2]
    const uint16_t* __restrict__ q,    // [B, Hq, D] rows at q_stride
    const void* __restrict__ k_cache_v,  // [NB, Hk, BS, D] bf16 | fp8
    const void* __restrict__ v_cache_v,  // [NB, Hk, D, BS] (transposed)
    const int* __restrict__ block_tables,
    const int* __restrict__ seq_lens,
    const int max_blocks, const int Hk, const int G, const int64_t q_stride,
    const float scale, const int sliding_window, const float softcap,
    const float* __restrict__ sinks, const int part_tokens,
    const float k_scale, const float v_scale,
    const int* __restrict__ token_positions = nullptr,  // [B(,Hk), max_pos]
    const int max_positions = 0, const int pos_heads = 1) {
  const int seq = blockIdx.y;
  const int hk = blockIdx.x;
  const int L = seq_lens[seq];
  const int* pos_row =
      SPARSE ? token_positions +
                   ((size_t)seq * pos_heads + (pos_heads > 1 ? hk : 0)) *
                       max_positions
             : nullptr;
  const int domain = SPARSE ? max_positions : L;

  int tok_begin = 0, tok_end = domain;
  if (!SPARSE && sliding_window > 0) tok_begin = max(0, L - sliding_window);
  if (PARTITIONED) {
    const int p = blockIdx.z;
    tok_begin = max(tok_begin, p * part_tokens);
    tok_end = min(tok_end, (p + 1) * part_tokens);
  }

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int l15 = lane & 15;
  const int l4 = lane >> 4;

  if (tok_begin >= tok_end) {
    if (PARTITIONED) {
      const int p = blockIdx.z;
      for (int g = tid; g < G; g += DM_THREADS) {
        float* ml =
            tmp_ml + (((size_t)seq * Hk * G + hk * G + g) * gridDim.z + p) * 2;
        ml[0] = -1e30f;
        ml[1] = 0.f;
      }
    }
    return;
  }

  __shared__ uint16_t Pl[16 * KT];  // row = q head, swz
  __shared__ float m_s[16], l_s[16], resc[16];
  __shared__ float wred[4][16];

  constexpr int KC = HEAD_DIM / 32;      // 16-B K/Q chunks per row
  constexpr int MT = HEAD_DIM / 16 / 4;  // PV M-tiles per wave
  constexpr int MS = KT / 64;            // token m-tiles per wave (QK)
  constexpr int SC = KT / 32;            // PV K-steps

  // ---- Q fragments in registers (head l15; zero-pad heads >= G) ---------------
  bf16x8v qfrag[KC];
  {
    const bool ok = l15 < G;
#pragma unroll
    for (int s = 0; s < KC; ++s) {
      int4 val = make_int4(0, 0, 0, 0);
      if (ok)
        val = *reinterpret_cast<const int4*>(
            q + (size_t)seq * q_stride + (hk * G + l15) * HEAD_DIM + s * 32 +
            l4 * 8);
      qfrag[s] = *reinterpret_cast<const bf16x8v*>(&val);
    }
  }
  if (tid < 16) {
    m_s[tid] = -3.0e4f;
    l_s[tid] = 0.f;
  }

  f32x4v acc_o[MT] = {};
  const int* btab = block_tables + (size_t)seq * max_blocks;
  const int kt_begin = tok_begin / KT;
  const int kt_end = (tok_end + KT - 1) / KT;

  // logical token -> global token (sparse indirection; clamped for address
  // safety — masked scores / zero p neutralize the values)
  auto map_tok = [&](int j, bool& valid) -> int {
    valid = j >= tok_begin && j < tok_end;
    int gt = j;
    if (SPARSE) {
      gt = valid ? pos_row[j] : 0;
      valid = valid && gt >= 0 && gt < L;
    }
    return min(max(gt, 0), L - 1);
  };

  // K: wave wid owns token rows [wid*16*MS, +16*MS); lane l15 picks the row
  // within each 16-row m-tile, all KC d-chunks per row.
  int4 kregA[MS][KC], kregB[MS][KC];
  // V^T: lane covers d rows (wid*MT+mt)*16 + l15, token chunk s*32 + l4*8.
  int4 vreg[SC][MT];

  auto load_k = [&](int kt, int4 (&kreg)[MS][KC]) {
#pragma unroll
    for (int ms = 0; ms < MS; ++ms) {
      bool valid;
      const int gtok =
          map_tok(kt * KT + (wid * MS + ms) * 16 + l15, valid);
      const int blk = btab[gtok / BLOCK_SIZE];
      const size_t row =
          (((size_t)blk * Hk + hk) * BLOCK_SIZE + gtok % BLOCK_SIZE) * HEAD_DIM;
#pragma unroll
      for (int s = 0; s < KC; ++s) {
        if (KV_FP8) {
          const uint64_t raw = *reinterpret_cast<const uint64_t*>(
              (const uint8_t*)k_cache_v + row + s * 32 + l4 * 8);
          fp8x8_to_bf16x8(raw, k_scale,
                          reinterpret_cast<uint16_t*>(&kreg[ms][s]));
        } else {
          kreg[ms][s] = *reinterpret_cast<const int4*>(
              (const uint16_t*)k_cache_v + row + s * 32 + l4 * 8);
        }
      }
    }
  };

  auto load_v = [&](int kt) {
#pragma unroll
    for (int s = 0; s < SC; ++s) {
      const int j0 = kt * KT + s * 32 + l4 * 8;
#pragma unroll
      for (int mt = 0; mt < MT; ++mt) {
        const int drow = (wid * MT + mt) * 16 + l15;
        if (SPARSE) {
          // sparse tokens are scattered: per-element gather
          alignas(16) uint16_t e[8];
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            bool valid;
            const int gt = map_tok(j0 + j, valid);
            const int blk = btab[gt / BLOCK_SIZE];
            const size_t addr = (((size_t)blk * Hk + hk) * HEAD_DIM + drow) *
                                    BLOCK_SIZE + gt % BLOCK_SIZE;
            if (KV_FP8)
              e[j] = f32_to_bf16_bits(
                  fp8_e4m3_to_f32(((const uint8_t*)v_cache_v)[addr]) * v_scale);
            else
              e[j] = ((const uint16_t*)v_cache_v)[addr];
          }
          vreg[s][mt] = *reinterpret_cast<const int4*>(e);
        } else {
          bool valid;
          const int gt = map_tok(j0, valid);  // 8 tokens stay in one block
          const int blk = btab[gt / BLOCK_SIZE];
          const size_t row = (((size_t)blk * Hk + hk) * HEAD_DIM + drow) *
                                 BLOCK_SIZE + gt % BLOCK_SIZE;
          if (KV_FP8) {
            const uint64_t raw = *reinterpret_cast<const uint64_t*>(
                (const uint8_t*)v_cache_v + row);
            fp8x8_to_bf16x8(raw, v_scale,
                            reinterpret_cast<uint16_t*>(&vreg[s][mt]));
          } else {
            vreg[s][mt] = *reinterpret_cast<const int4*>(
                (const uint16_t*)v_cache_v + row);
          }
        }
      }
    }
  };

  auto tile_math = [&](int kt, const int4 (&kreg)[MS][KC]) {
    const int kbase = kt * KT;

    // ---- S^T = K . Q^T (wave w: token rows [w*16*MS, +16*MS)) -------------------
    f32x4v acc_s[MS] = {};
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ms = 0; ms < MS; ++ms)
#pragma unroll
      for (int s = 0; s < KC; ++s) {
        const bf16x8v afrag = *reinterpret_cast<const bf16x8v*>(&kreg[ms][s]);
        acc_s[ms] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afrag, qfrag[s], acc_s[ms], 0, 0, 0);
      }
    __builtin_amdgcn_s_setprio(0);

    // ---- mask + per-column max --------------------------------------------------
    float mx = -3.0e4f;
#pragma unroll
    for (int ms = 0; ms < MS; ++ms)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int j = kbase + (wid * MS + ms) * 16 + l4 * 4 + r;
        float s = acc_s[ms][r] * scale;
        if (softcap > 0.f) s = softcap * tanhf(s / softcap);
        bool visible = j >= tok_begin && j < tok_end && l15 < G;
        if (SPARSE && visible) {
          const int gt = pos_row[j];
          visible = gt >= 0 && gt < L;
        }
        s = visible ? s : -3.0e4f;
        acc_s[ms][r] = s;
        mx = fmaxf(mx, s);
      }
    mx = fmaxf(mx, __shfl_xor(mx, 16, WAVE_SIZE));
    mx = fmaxf(mx, __shfl_xor(mx, 32, WAVE_SIZE));
    if (lane < 16) wred[wid][lane] = mx;
    __syncthreads();
    if (tid < 16) {
      const float m_chunk = fmaxf(fmaxf(wred[0][tid], wred[1][tid]),
                                  fmaxf(wred[2][tid], wred[3][tid]));
      const float m_new = fmaxf(m_s[tid], m_chunk);
      resc[tid] = __expf(m_s[tid] - m_new);
      m_s[tid] = m_new;
    }
    __syncthreads();

    // ---- p = exp(s - m) -> Pl; column sums ---------------------------------------
    float sm = 0.f;
    {
      const float m = m_s[l15];
#pragma unroll
      for (int ms = 0; ms < MS; ++ms)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const float p = __expf(acc_s[ms][r] - m);
          sm += p;
          const int kk = (wid * MS + ms) * 16 + l4 * 4 + r;
          const int pb = swz(l15 * KT * 2 + kk * 2, l15);
          *reinterpret_cast<uint16_t*>(reinterpret_cast<char*>(Pl) + pb) =
              f32_to_bf16_bits(p);
        }
      sm += __shfl_xor(sm, 16, WAVE_SIZE);
      sm += __shfl_xor(sm, 32, WAVE_SIZE);
    }
    if (lane < 16) wred[wid][lane] = sm;
    __syncthreads();  // Pl + wred writes from every wave before l_s / PV
    if (tid < 16)
      l_s[tid] = l_s[tid] * resc[tid] + wred[0][tid] + wred[1][tid] +
                 wred[2][tid] + wred[3][tid];

    // ---- O^T += V^T . P ----------------------------------------------------------
    {
      const float r = resc[l15];
#pragma unroll
      for (int mt = 0; mt < MT; ++mt)
#pragma unroll
        for (int rr = 0; rr < 4; ++rr) acc_o[mt][rr] *= r;
    }
    // barrier: l_s's wred read + everyone's resc read must complete before
    // the next tile's wred/resc writes (threads race ahead through PV)
    __syncthreads();
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int s = 0; s < SC; ++s) {
      const int pb = swz(l15 * KT * 2 + s * 64 + l4 * 16, l15);
      const bf16x8v bfrag = *reinterpret_cast<const bf16x8v*>(
          reinterpret_cast<const char*>(Pl) + pb);
#pragma unroll
      for (int mt = 0; mt < MT; ++mt) {
        const bf16x8v afrag = *reinterpret_cast<const bf16x8v*>(&vreg[s][mt]);
        acc_o[mt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag,
                                                            acc_o[mt], 0, 0, 0);
      }
    }
    __builtin_amdgcn_s_setprio(0);
    // no trailing barrier: the next tile's two reduction barriers precede any
    // Pl/resc rewrite, so this tile's PV reads are already protected
  };

  load_k(kt_begin, kregA);
  for (int kt = kt_begin; kt < kt_end; ++kt) {
    // V for THIS tile issues first (its latency hides under QK+softmax);
    // next tile's K prefetches into the other named buffer.
    load_v(kt);
    if (((kt - kt_begin) & 1) == 0) {
      if (kt + 1 < kt_end) load_k(kt + 1, kregB);
      tile_math(kt, kregA);
    } else {
      if (kt + 1 < kt_end) load_k(kt + 1, kregA);
      tile_math(kt, kregB);
    }
  }

  // ---- write out (O^T acc: col=q head, row=d) -----------------------------------
  const int g = l15;
  if (g < G) {
    const size_t hq = (size_t)hk * G + g;
    if (PARTITIONED) {
      const int p = blockIdx.z;
#pragma unroll
      for (int mt = 0; mt < MT; ++mt) {
        const int d0 = (wid * MT + mt) * 16 + l4 * 4;
        float* dst =
            tmp_acc + (((size_t)seq * Hk * G + hq) * gridDim.z + p) * HEAD_DIM + d0;
#pragma unroll
        for (int r = 0; r < 4; ++r) dst[r] = acc_o[mt][r];
      }
      if (wid == 0 && l4 == 0) {
        float* ml =
            tmp_ml + (((size_t)seq * Hk * G + hq) * gridDim.z + blockIdx.z) * 2;
        ml[0] = m_s[g];
        ml[1] = l_s[g];
      }
    } else {
      float l = l_s[g];
      if (sinks != nullptr) l += __expf(sinks[hq] - m_s[g]);
      const float inv = 1.f / l;
#pragma unroll
      for (int mt = 0; mt < MT; ++mt) {
        const int d0 = (wid * MT + mt) * 16 + l4 * 4;
        uint16_t vals[4];
#pragma unroll
        for (int r = 0; r < 4; ++r)
          vals[r] = f32_to_bf16_bits(acc_o[mt][r] * inv);
        *reinterpret_cast<uint2*>(out + hq * HEAD_DIM +
                                  (size_t)seq * Hk * G * HEAD_DIM + d0) =
            *reinterpret_cast<const uint2*>(vals);
      }
    }
  }
}

static int dm_ktile(int span_tokens) {
  // A/B on MI355X (profiles/README.md round 2): KT=128 wins when each
  // workgroup walks a long token span (ctx 8k: 5.63 vs 5.08 TB/s), KT=64
  // wins on short spans (less boundary-tile waste). Env overrides for A/B.
  static const int forced = [] {
    const char* e = getenv("PARALLAX_DM_KTILE");
    return e ? atoi(e) : 0;
  }();
  if (forced == 64 || forced == 128) return forced;
  return span_tokens >= 2048 ? 128 : 64;
}

extern "C" void launch_msa_paged_attention_decode(
    void* out, const void* q, const void* k_cache, const void* v_cache,
    const int* block_tables, const int* seq_lens, const int* token_positions,
    int max_positions, int pos_heads, int B, int Hq, int Hk, int D, int BS,
    int max_blocks, int64_t q_stride, float scale, hipStream_t stream,
    bool* launched) {
  const int G = Hq / Hk;
  *launched = false;
  if (G > 16) return;
  dim3 grid(Hk, B, 1);
  // sparse tiles gather per-element anyway: KT=64 keeps the tail waste low
#define MSA_LAUNCH(HD, BSZ)                                                    \
  paged_decode_mfma_kernel<HD, BSZ, false, false, true, 64>                    \
      <<<grid, DM_THREADS, 0, stream>>>(                                       \
          (uint16_t*)out, nullptr, nullptr, (const uint16_t*)q, k_cache,       \
          v_cache, block_tables, seq_lens, max_blocks, Hk, G, q_stride, scale, \
          -1, 0.f, nullptr, 0, 1.f, 1.f, token_positions, max_positions,      \
          pos_heads);                                                          \
  *launched = true;
  if (D == 128 && BS == 32) { MSA_LAUNCH(128, 32) }
  else if (D == 128 && BS == 16) { MSA_LAUNCH(128, 16) }
  else if (D == 128 && BS == 64) { MSA_LAUNCH(128, 64) }
  else if (D == 64 && BS == 32) { MSA_LAUNCH(64, 32) }
  else if (D == 64 && BS == 16) { MSA_LAUNCH(64, 16) }
  else if (D == 64 && BS == 64) { MSA_LAUNCH(64, 64) }
#undef MSA_LAUNCH
}

extern "C" void launch_paged_decode_mfma(
    void* out, const void* q, const void* k_cache, const void* v_cache,
    const int* block_tables, const int* seq_lens, int B, int Hq, int Hk, int D,
    int BS, int max_blocks, int64_t q_stride, float scale, int sliding_window,
    float softcap, const float* sinks, int num_parts, int part_tokens,
    float* tmp_acc, float* tmp_ml, bool kv_fp8, float k_scale, float v_scale,
    hipStream_t stream, bool* launched) {
  const int G = Hq / Hk;
  *launched = false;
  if (G > 16) return;

#define DM_LAUNCH3(HD, BSZ, FP8, KTV)                                          \
  if (num_parts <= 1) {                                                        \
    dim3 grid(Hk, B, 1);                                                       \
    paged_decode_mfma_kernel<HD, BSZ, false, FP8, false, KTV>                  \
        <<<grid, DM_THREADS, 0, stream>>>(                                     \
            (uint16_t*)out, nullptr, nullptr, (const uint16_t*)q, k_cache,     \
            v_cache, block_tables, seq_lens, max_blocks, Hk, G, q_stride,      \
            scale, sliding_window, softcap, sinks, 0, k_scale, v_scale);       \
  } else {                                                                     \
    dim3 grid(Hk, B, num_parts);                                               \
    paged_decode_mfma_kernel<HD, BSZ, true, FP8, false, KTV>                   \
        <<<grid, DM_THREADS, 0, stream>>>(                                     \
            nullptr, tmp_acc, tmp_ml, (const uint16_t*)q, k_cache, v_cache,    \
            block_tables, seq_lens, max_blocks, Hk, G, q_stride, scale,        \
            sliding_window, softcap, sinks, part_tokens, k_scale, v_scale);    \
  }                                                                            \
  *launched = true;

  const int span = num_parts > 1 ? part_tokens : 1 << 30;
  const int kt_pick = dm_ktile(span);

#define DM_LAUNCH(HD, BSZ)                                                     \
  if (kv_fp8) {                                                                \
    if (kt_pick == 128) { DM_LAUNCH3(HD, BSZ, true, 128) }                     \
    else { DM_LAUNCH3(HD, BSZ, true, 64) }                                     \
  } else {                                                                     \
    if (kt_pick == 128) { DM_LAUNCH3(HD, BSZ, false, 128) }                    \
    else { DM_LAUNCH3(HD, BSZ, false, 64) }                                    \
  }

  if (D == 128 && BS == 32) { DM_LAUNCH(128, 32) }
  else if (D == 128 && BS == 16) { DM_LAUNCH(128, 16) }
  else if (D == 128 && BS == 64) { DM_LAUNCH(128, 64) }
  else if (D == 64 && BS == 32) { DM_LAUNCH(64, 32) }
  else if (D == 64 && BS == 16) { DM_LAUNCH(64, 16) }
  else if (D == 64 && BS == 64) { DM_LAUNCH(64, 64) }
#undef DM_LAUNCH
#undef DM_LAUNCH3
}
