// DeepSeek MLA decode attention over the compressed latent cache — MFMA/CDNA4.
//
//   S^T[32tok x 32heads] = C_tile · Q^T   (C = [latent | rope] rows, 576 dims;
//                                          Q fragments read straight from
//                                          q_latent/q_pe global — L1-resident)
//   O^T[512 x 32heads]  += C_v^T · P      (latent transposed during staging)
//
// All query heads share ONE latent cache (MQA-like), so arithmetic intensity is
// ~240 flop/byte — compute-bound, hence matrix cores. A workgroup covers a
// 32-head block; the cache tile re-read across the H/32 head-block workgroups
// of a sequence is L2/L3-absorbed. Fixed-size context partitions (graph-stable)
// combine through the shared reduce kernel.
//
// Reference analogue: parallax_extensions/kernels/mla/ (Metal, VALU); fresh
// MFMA design. Layout: cache [NB, BS, R+DR], q_latent [B, H, R], q_pe [B, H, DR].

#include "common.h"

#define MLA_THREADS 256
#define MLA_KTILE 32
#define MLA_HBLOCK 32

template <int R, int DR, int BLOCK_SIZE, bool PARTITIONED, bool SPARSE,
          bool KV_FP8 = false>
__global__ __launch_bounds__(MLA_THREADS) void mla_decode_kernel(
    uint16_t* __restrict__ out,       // [B, H, R]
    float* __restrict__ tmp_acc,      // [B, H, P, R]
    float* __restrict__ tmp_ml,      // [B, H, P, 2]
    const uint16_t* __restrict__ q_latent,  // [B, H, R]
    const uint16_t* __restrict__ q_pe,      // [B, H, DR]
    const void* __restrict__ cache_v,       // [NB, BS, R+DR] bf16 | fp8(e4m3)
    const int* __restrict__ block_tables,
    const int* __restrict__ seq_lens,
    const int max_blocks, const int H, const float scale,
    const int part_tokens,
    const int* __restrict__ topk_indices,  // [B, max_topk] (SPARSE only)
    const int max_topk, const float c_scale = 1.f) {
  constexpr int DK = R + DR;            // 576
  constexpr int KSTEPS = DK / 32;       // 18
  constexpr int PROW = 40;              // padded row length (bank spread)
  const int hb = blockIdx.x;            // head block (32 heads)
  const int seq = blockIdx.y;
  const int L = seq_lens[seq];
  // DSA sparse mode: iterate the top-k index list; a row starting with -1
  // falls back to dense (reference dsa_paged_attention semantics)
  const int* idx_row = SPARSE ? topk_indices + (size_t)seq * max_topk : nullptr;
  const bool sparse_row = SPARSE && idx_row[0] >= 0;
  const int domain = sparse_row ? max_topk : L;

  int tok_begin = 0, tok_end = domain;
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
      for (int i = tid; i < MLA_HBLOCK; i += MLA_THREADS) {
        const int h = hb * MLA_HBLOCK + i;
        if (h < H) {
          float* ml = tmp_ml + (((size_t)seq * H + h) * gridDim.z + p) * 2;
          ml[0] = -1e30f;
          ml[1] = 0.f;
        }
      }
    }
    return;
  }

  __shared__ uint16_t Cl[MLA_KTILE * DK];        // row = token (1152 B), swz
  __shared__ uint16_t VTl[R * PROW];             // row = latent dim, 80 B pad
  __shared__ uint16_t Pl[MLA_HBLOCK * PROW];     // row = head, 80 B pad
  __shared__ float m_s[MLA_HBLOCK], l_s[MLA_HBLOCK], resc[MLA_HBLOCK];
  __shared__ float wred[4][MLA_HBLOCK];

  if (tid < MLA_HBLOCK) {
    m_s[tid] = -3.0e4f;
    l_s[tid] = 0.f;
  }

  // O^T acc: wave w owns latent rows [128w, 128w+128) = 8 M-tiles x 2 N-tiles
  constexpr int MT = R / 16 / 4;  // 8
  f32x4v acc_o[MT][2] = {};

  const int* btab = block_tables + (size_t)seq * max_blocks;
  const int kt_begin = tok_begin / MLA_KTILE;
  const int kt_end = (tok_end + MLA_KTILE - 1) / MLA_KTILE;

  for (int kt = kt_begin; kt < kt_end; ++kt) {
    const int kbase = kt * MLA_KTILE;
    __syncthreads();

    // ---- stage C rows (swz) + transposed latent (padded rows) ------------------
    {
      const int tok = tid & 31;
      const int dv = tid >> 5;  // 8 slices of 72 dims (9 int4 each)
      const int j = kbase + tok;
      int gtok = j;
      bool ok = j >= tok_begin && j < tok_end;
      if (sparse_row && ok) {
        gtok = idx_row[j];
        ok = gtok >= 0 && gtok < L;
      }
      size_t crow_tok = 0;
      if (ok) {
        const int blk = btab[gtok / BLOCK_SIZE];
        const int off = gtok % BLOCK_SIZE;
        crow_tok = (size_t)blk * BLOCK_SIZE + off;
      }
#pragma unroll
      for (int c = 0; c < 9; ++c) {
        const int d = (dv * 9 + c) * 8;
        int4 val = make_int4(0, 0, 0, 0);
        if (ok) {
          if (KV_FP8) {
            // 8 fp8 bytes -> 8 bf16 (one cache-wide scale), packed converts
            const uint64_t raw = *reinterpret_cast<const uint64_t*>(
                (const uint8_t*)cache_v + crow_tok * DK + d);
            fp8x8_to_bf16x8(raw, c_scale, reinterpret_cast<uint16_t*>(&val));
          } else {
            val = *reinterpret_cast<const int4*>(
                (const uint16_t*)cache_v + crow_tok * DK + d);
          }
        }
        const int byte = swz(tok * DK * 2 + d * 2, tok);
        *reinterpret_cast<int4*>(reinterpret_cast<char*>(Cl) + byte) = val;
        const uint16_t* vs = reinterpret_cast<const uint16_t*>(&val);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const int dd = d + j;
          if (dd < R)
            VTl[dd * PROW + tok] = vs[j];
        }
      }
    }
    __syncthreads();

    // ---- S^T[32tok x 32h] = C . Q^T ---------------------------------------------
    // waves: w>>1 selects the 16-token M strip, w&1 selects the 16-head N strip
    const int mw = wid >> 1;  // 0..1
    const int nw = wid & 1;   // 0..1
    f32x4v acc_s = {};
    const int head = hb * MLA_HBLOCK + nw * 16 + l15;
    const bool head_ok = head < H;
#pragma unroll
    for (int s = 0; s < KSTEPS; ++s) {
      const int trow = mw * 16 + l15;
      const int ca = swz(trow * DK * 2 + s * 64 + l4 * 16, trow);
      const bf16x8v afrag = *reinterpret_cast<const bf16x8v*>(
          reinterpret_cast<const char*>(Cl) + ca);
      bf16x8v bfrag = {};
      if (head_ok) {
        const int k = s * 32 + l4 * 8;
        if (k < R)
          bfrag = *reinterpret_cast<const bf16x8v*>(
              q_latent + ((size_t)seq * H + head) * R + k);
        else
          bfrag = *reinterpret_cast<const bf16x8v*>(
              q_pe + ((size_t)seq * H + head) * DR + (k - R));
      }
      acc_s = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc_s, 0, 0, 0);
    }

    // ---- mask + softmax (per head column, online) ------------------------------
    float mx = -3.0e4f;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int j = kbase + mw * 16 + l4 * 4 + r;
      float s = acc_s[r] * scale;
      bool vis = j >= tok_begin && j < tok_end && head_ok;
      if (sparse_row && vis) {
        const int gt = idx_row[j];
        vis = gt >= 0 && gt < L;
      }
      s = vis ? s : -3.0e4f;
      acc_s[r] = s;
      mx = fmaxf(mx, s);
    }
    // same head column lives in lanes {c, c+16, c+32, c+48} of this wave
    mx = fmaxf(mx, __shfl_xor(mx, 16, WAVE_SIZE));
    mx = fmaxf(mx, __shfl_xor(mx, 32, WAVE_SIZE));
    if (lane < 16) wred[wid][nw * 16 + lane] = mx;
    __syncthreads();
    if (tid < MLA_HBLOCK) {
      // waves (0,2) cover N strip 0; (1,3) N strip 1 — combine the two M strips
      const int nwi = tid >> 4;
      const float m_chunk = fmaxf(wred[nwi][tid], wred[nwi + 2][tid]);
      const float m_new = fmaxf(m_s[tid], m_chunk);
      resc[tid] = __expf(m_s[tid] - m_new);
      m_s[tid] = m_new;
    }
    __syncthreads();

    float sm = 0.f;
    {
      const float m = m_s[nw * 16 + l15];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const float p = __expf(acc_s[r] - m);
        sm += p;
        const int kk = mw * 16 + l4 * 4 + r;
        Pl[(nw * 16 + l15) * PROW + kk] = f32_to_bf16_bits(p);
      }
      sm += __shfl_xor(sm, 16, WAVE_SIZE);
      sm += __shfl_xor(sm, 32, WAVE_SIZE);
    }
    if (lane < 16) wred[wid][nw * 16 + lane] = sm;
    __syncthreads();
    if (tid < MLA_HBLOCK) {
      const int nwi = tid >> 4;
      l_s[tid] = l_s[tid] * resc[tid] + wred[nwi][tid] + wred[nwi + 2][tid];
    }
    __syncthreads();

    // ---- O^T += C_v^T . P --------------------------------------------------------
#pragma unroll
    for (int mt = 0; mt < MT; ++mt)
#pragma unroll
      for (int nt = 0; nt < 2; ++nt) {
        const float rsc = resc[nt * 16 + l15];
#pragma unroll
        for (int rr = 0; rr < 4; ++rr) acc_o[mt][nt][rr] *= rsc;
      }
#pragma unroll
    for (int mt = 0; mt < MT; ++mt) {
      const int rrow = wid * (R / 4) + mt * 16 + l15;
      const bf16x8v afrag = *reinterpret_cast<const bf16x8v*>(
          VTl + rrow * PROW + l4 * 8);
#pragma unroll
      for (int nt = 0; nt < 2; ++nt) {
        const bf16x8v bfrag = *reinterpret_cast<const bf16x8v*>(
            Pl + (nt * 16 + l15) * PROW + l4 * 8);
        acc_o[mt][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afrag, bfrag, acc_o[mt][nt], 0, 0, 0);
      }
    }
  }

  // ---- write out (O^T: col = head, row = latent dim) ------------------------------
  __syncthreads();
#pragma unroll
  for (int nt = 0; nt < 2; ++nt) {
    const int h = hb * MLA_HBLOCK + nt * 16 + l15;
    if (h >= H) continue;
    if (PARTITIONED) {
      const int p = blockIdx.z;
#pragma unroll
      for (int mt = 0; mt < MT; ++mt) {
        const int r0 = wid * (R / 4) + mt * 16 + l4 * 4;
        float* dst = tmp_acc + (((size_t)seq * H + h) * gridDim.z + p) * R + r0;
#pragma unroll
        for (int rr = 0; rr < 4; ++rr) dst[rr] = acc_o[mt][nt][rr];
      }
      if (wid == 0 && l4 == 0) {
        float* ml = tmp_ml + (((size_t)seq * H + h) * gridDim.z + blockIdx.z) * 2;
        ml[0] = m_s[nt * 16 + l15];
        ml[1] = l_s[nt * 16 + l15];
      }
    } else {
      const float inv = 1.f / l_s[nt * 16 + l15];
#pragma unroll
      for (int mt = 0; mt < MT; ++mt) {
        const int r0 = wid * (R / 4) + mt * 16 + l4 * 4;
        uint16_t vals[4];
#pragma unroll
        for (int rr = 0; rr < 4; ++rr)
          vals[rr] = f32_to_bf16_bits(acc_o[mt][nt][rr] * inv);
        *reinterpret_cast<uint2*>(out + ((size_t)seq * H + h) * R + r0) =
            *reinterpret_cast<const uint2*>(vals);
      }
    }
  }
}

extern "C" void launch_mla_paged_attention_decode(
    void* out, const void* q_latent, const void* q_pe, const void* cache,
    const int* block_tables, const int* seq_lens, int B, int H, int R, int DR,
    int BS, int max_blocks, float scale, int num_parts, int part_tokens,
    float* tmp_acc, float* tmp_ml, const int* topk_indices, int max_topk,
    bool kv_fp8, float c_scale, hipStream_t stream, bool* launched) {
  *launched = false;
  if (R != 512 || DR != 64) return;  // DeepSeek V2/V3/R1/K2 geometry
  const int head_blocks = (H + MLA_HBLOCK - 1) / MLA_HBLOCK;

#define MLA_LAUNCH3(BSZ, SP, FP8)                                             \
  if (num_parts <= 1) {                                                       \
    dim3 grid(head_blocks, B, 1);                                             \
    mla_decode_kernel<512, 64, BSZ, false, SP, FP8>                           \
        <<<grid, MLA_THREADS, 0, stream>>>(                                   \
        (uint16_t*)out, nullptr, nullptr, (const uint16_t*)q_latent,          \
        (const uint16_t*)q_pe, cache, block_tables,                           \
        seq_lens, max_blocks, H, scale, 0, topk_indices, max_topk, c_scale);  \
  } else {                                                                    \
    dim3 grid(head_blocks, B, num_parts);                                     \
    mla_decode_kernel<512, 64, BSZ, true, SP, FP8>                            \
        <<<grid, MLA_THREADS, 0, stream>>>(                                   \
        nullptr, tmp_acc, tmp_ml, (const uint16_t*)q_latent,                  \
        (const uint16_t*)q_pe, cache, block_tables,                           \
        seq_lens, max_blocks, H, scale, part_tokens, topk_indices, max_topk,  \
        c_scale);                                                             \
  }                                                                           \
  *launched = true;

#define MLA_LAUNCH2(BSZ, SP)                                                  \
  if (kv_fp8) { MLA_LAUNCH3(BSZ, SP, true) }                                  \
  else { MLA_LAUNCH3(BSZ, SP, false) }

#define MLA_LAUNCH(BSZ)                                                       \
  if (topk_indices != nullptr) { MLA_LAUNCH2(BSZ, true) }                     \
  else { MLA_LAUNCH2(BSZ, false) }

  if (BS == 32) { MLA_LAUNCH(32) }
  else if (BS == 16) { MLA_LAUNCH(16) }
  else if (BS == 64) { MLA_LAUNCH(64) }
#undef MLA_LAUNCH
#undef MLA_LAUNCH2
#undef MLA_LAUNCH3
}
