#include "hip/hip_runtime.h"
// MFMA decode attention: the GQA query group rides the matrix cores.
//
// The VALU decode kernel (paged_attention.hip) computes G x D dot products per
// token with unpack+fma chains — fine at G<=4, VALU-bound at G>=8. Here the
// whole KV group is one MFMA N-tile (G <= 16 query heads = 16 columns):
//   S^T[64tok x 16q] = K_tile · Q^T      (A = K rows, B = Q rows — natural)
//   O^T[D x 16q]    += V^T · P           (V transposed during staging)
// using the same LDS layout machinery as the prefill kernel (XOR-swizzled rows,
// ds_read_b128 fragments). One workgroup per (sequence, kv_head, partition);
// partitions are fixed-size (graph-capture stable) and combine through the
// same reduce kernel as the VALU path.

#include "common.h"

#define DM_THREADS 256
#define DM_KTILE 64

template <int HEAD_DIM, int BLOCK_SIZE, bool PARTITIONED, bool KV_FP8,
          bool SPARSE = false>
__global__ __launch_bounds__(DM_THREADS) void paged_decode_mfma_kernel(
    uint16_t* __restrict__ out,        // [B, Hq, D] (final mode)
    float* __restrict__ tmp_acc,       // [B, Hq, P, D] (partitioned)
    float* __restrict__ tmp_ml,        // [B, Hq, P, 2]
    const uint16_t* __restrict__ q,    // [B, Hq, D] rows at q_stride
    const void* __restrict__ k_cache_v,  // [NB, Hk, BS, D] bf16 | fp8
    const void* __restrict__ v_cache_v,
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
  // MSA sparse mode: iterate an explicit token-position list (-1 padded);
  // pos_heads > 1 = one independent selection per kv head (minimax-m3)
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

  __shared__ uint16_t Kl[DM_KTILE * HEAD_DIM];   // row = token, swz
  __shared__ uint16_t VTl[HEAD_DIM * DM_KTILE];  // row = dim,   swz
  __shared__ uint16_t Ql[16 * HEAD_DIM];         // row = q head, swz
  __shared__ uint16_t Pl[16 * DM_KTILE];         // row = q head, swz
  __shared__ float m_s[16], l_s[16], resc[16];
  __shared__ float wred[4][16];

  // ---- stage the G query heads (zeros pad to 16 rows) ---------------------------
  {
    const int qrow = tid & 15;
    const int dv = tid >> 4;  // 16 chunks
    const bool ok = qrow < G;
#pragma unroll
    for (int c = 0; c < HEAD_DIM / 128; ++c) {
      const int d = (dv + c * 16) * 8;
      int4 val = make_int4(0, 0, 0, 0);
      if (ok)
        val = *reinterpret_cast<const int4*>(
            q + (size_t)seq * q_stride + (hk * G + qrow) * HEAD_DIM + d);
      const int byte = swz(qrow * HEAD_DIM * 2 + d * 2, qrow);
      *reinterpret_cast<int4*>(reinterpret_cast<char*>(Ql) + byte) = val;
    }
    if (HEAD_DIM == 64 && dv < 8) {  // D=64: only 8 chunks
      const int d = dv * 8;
      int4 val = make_int4(0, 0, 0, 0);
      if (ok)
        val = *reinterpret_cast<const int4*>(
            q + (size_t)seq * q_stride + (hk * G + qrow) * HEAD_DIM + d);
      const int byte = swz(qrow * HEAD_DIM * 2 + d * 2, qrow);
      *reinterpret_cast<int4*>(reinterpret_cast<char*>(Ql) + byte) = val;
    }
  }
  if (tid < 16) {
    m_s[tid] = -3.0e4f;
    l_s[tid] = 0.f;
  }

  constexpr int MT = HEAD_DIM / 16 / 4;  // M-tiles of O^T per wave
  f32x4v acc_o[MT] = {};

  const int* btab = block_tables + (size_t)seq * max_blocks;
  const int kt_begin = tok_begin / DM_KTILE;
  const int kt_end = (tok_end + DM_KTILE - 1) / DM_KTILE;

  // register double-buffered staging (async-STAGE split): loads for tile n+1
  // are issued right after tile n's LDS write so HBM latency hides under the
  // MFMA/softmax work (guide §6 G15)
  const int stg_tok = tid & 63;
  const int stg_dv = tid >> 6;
  const int stg_d0 = stg_dv * (HEAD_DIM / 4);
  int4 kreg[HEAD_DIM / 32], vreg[HEAD_DIM / 32];

  auto load_tile = [&](int kt) {
    const int j = kt * DM_KTILE + stg_tok;
    int gtok = j;
    bool ok = j >= tok_begin && j < tok_end;
    if (SPARSE && ok) {
      gtok = pos_row[j];
      ok = gtok >= 0 && gtok < L;
    }
    size_t row_off = 0;
    if (ok) {
      const int blk = btab[gtok / BLOCK_SIZE];
      const int off = gtok % BLOCK_SIZE;
      row_off = (((size_t)blk * Hk + hk) * BLOCK_SIZE + off) * HEAD_DIM;
    }
#pragma unroll
    for (int c = 0; c < HEAD_DIM / 32; ++c) {
      const int d = stg_d0 + c * 8;
      int4 kval = make_int4(0, 0, 0, 0);
      int4 vval = make_int4(0, 0, 0, 0);
      if (ok) {
        if (KV_FP8) {
          // 8 fp8 bytes -> 8 bf16 (scale folded in)
          const uint64_t kraw = *reinterpret_cast<const uint64_t*>(
              (const uint8_t*)k_cache_v + row_off + d);
          const uint64_t vraw = *reinterpret_cast<const uint64_t*>(
              (const uint8_t*)v_cache_v + row_off + d);
          uint16_t* ks = reinterpret_cast<uint16_t*>(&kval);
          uint16_t* vsp = reinterpret_cast<uint16_t*>(&vval);
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            ks[j] = f32_to_bf16_bits(
                fp8_e4m3_to_f32((kraw >> (8 * j)) & 0xff) * k_scale);
            vsp[j] = f32_to_bf16_bits(
                fp8_e4m3_to_f32((vraw >> (8 * j)) & 0xff) * v_scale);
          }
        } else {
          kval = *reinterpret_cast<const int4*>(
              (const uint16_t*)k_cache_v + row_off + d);
          vval = *reinterpret_cast<const int4*>(
              (const uint16_t*)v_cache_v + row_off + d);
        }
      }
      kreg[c] = kval;
      vreg[c] = vval;
    }
  };

  load_tile(kt_begin);

  for (int kt = kt_begin; kt < kt_end; ++kt) {
    const int kbase = kt * DM_KTILE;
    __syncthreads();  // previous tile's MFMA done reading LDS

    // ---- write the prefetched tile: K row-major swz, V transposed ---------------
#pragma unroll
    for (int c = 0; c < HEAD_DIM / 32; ++c) {
      const int d = stg_d0 + c * 8;
      const int kb = swz(stg_tok * HEAD_DIM * 2 + d * 2, stg_tok);
      *reinterpret_cast<int4*>(reinterpret_cast<char*>(Kl) + kb) = kreg[c];
      const uint16_t* vs = reinterpret_cast<const uint16_t*>(&vreg[c]);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int dd = d + j;
        const int vb = swz(dd * DM_KTILE * 2 + stg_tok * 2, dd);
        *reinterpret_cast<uint16_t*>(reinterpret_cast<char*>(VTl) + vb) = vs[j];
      }
    }
    __syncthreads();
    if (kt + 1 < kt_end) load_tile(kt + 1);  // in flight during the math below

    // ---- S^T = K . Q^T (wave w: k rows [16w, 16w+16)) ---------------------------
    f32x4v acc_s = {};
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int s = 0; s < HEAD_DIM / 32; ++s) {
      const int krow_i = 16 * wid + l15;
      const int ka = swz(krow_i * HEAD_DIM * 2 + s * 64 + l4 * 16, krow_i);
      const bf16x8v afrag = *reinterpret_cast<const bf16x8v*>(
          reinterpret_cast<const char*>(Kl) + ka);
      const int qb = swz(l15 * HEAD_DIM * 2 + s * 64 + l4 * 16, l15);
      const bf16x8v bfrag = *reinterpret_cast<const bf16x8v*>(
          reinterpret_cast<const char*>(Ql) + qb);
      acc_s = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc_s, 0, 0, 0);
    }
    __builtin_amdgcn_s_setprio(0);

    // ---- mask + per-column max --------------------------------------------------
    float mx = -3.0e4f;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int j = kbase + 16 * wid + l4 * 4 + r;
      float s = acc_s[r] * scale;
      if (softcap > 0.f) s = softcap * tanhf(s / softcap);
      bool visible = j >= tok_begin && j < tok_end && l15 < G;
      if (SPARSE && visible) {
        const int gt = pos_row[j];
        visible = gt >= 0 && gt < L;
      }
      s = visible ? s : -3.0e4f;
      acc_s[r] = s;
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
      for (int r = 0; r < 4; ++r) {
        const float p = __expf(acc_s[r] - m);
        sm += p;
        const int kk = 16 * wid + l4 * 4 + r;
        const int pb = swz(l15 * DM_KTILE * 2 + kk * 2, l15);
        *reinterpret_cast<uint16_t*>(reinterpret_cast<char*>(Pl) + pb) =
            f32_to_bf16_bits(p);
      }
      sm += __shfl_xor(sm, 16, WAVE_SIZE);
      sm += __shfl_xor(sm, 32, WAVE_SIZE);
    }
    if (lane < 16) wred[wid][lane] = sm;
    __syncthreads();
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
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int s = 0; s < DM_KTILE / 32; ++s) {
      const int pb = swz(l15 * DM_KTILE * 2 + s * 64 + l4 * 16, l15);
      const bf16x8v bfrag = *reinterpret_cast<const bf16x8v*>(
          reinterpret_cast<const char*>(Pl) + pb);
#pragma unroll
      for (int mt = 0; mt < MT; ++mt) {
        const int drow = (wid * MT + mt) * 16 + l15;
        const int va = swz(drow * DM_KTILE * 2 + s * 64 + l4 * 16, drow);
        const bf16x8v afrag = *reinterpret_cast<const bf16x8v*>(
            reinterpret_cast<const char*>(VTl) + va);
        acc_o[mt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag,
                                                            acc_o[mt], 0, 0, 0);
      }
    }
    __builtin_amdgcn_s_setprio(0);
  }

  // ---- write out (O^T acc: col=q head, row=d) -----------------------------------
  __syncthreads();
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
#define MSA_LAUNCH(HD, BSZ)                                                    \
 hipLaunchKernelGGL(( paged_decode_mfma_kernel<HD, BSZ, false, false, true>)                        \
      , dim3(grid), dim3(DM_THREADS), 0, stream,                                        \
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

#define DM_LAUNCH2(HD, BSZ, FP8)                                               \
  if (num_parts <= 1) {                                                        \
    dim3 grid(Hk, B, 1);                                                       \
   hipLaunchKernelGGL(( paged_decode_mfma_kernel<HD, BSZ, false, FP8>)                              \
        , dim3(grid), dim3(DM_THREADS), 0, stream,                                      \
            (uint16_t*)out, nullptr, nullptr, (const uint16_t*)q, k_cache,     \
            v_cache, block_tables, seq_lens, max_blocks, Hk, G, q_stride,      \
            scale, sliding_window, softcap, sinks, 0, k_scale, v_scale);       \
  } else {                                                                     \
    dim3 grid(Hk, B, num_parts);                                               \
   hipLaunchKernelGGL(( paged_decode_mfma_kernel<HD, BSZ, true, FP8>)                               \
        , dim3(grid), dim3(DM_THREADS), 0, stream,                                      \
            nullptr, tmp_acc, tmp_ml, (const uint16_t*)q, k_cache, v_cache,    \
            block_tables, seq_lens, max_blocks, Hk, G, q_stride, scale,        \
            sliding_window, softcap, sinks, part_tokens, k_scale, v_scale);    \
  }                                                                            \
  *launched = true;

#define DM_LAUNCH(HD, BSZ)                                                     \
  if (kv_fp8) { DM_LAUNCH2(HD, BSZ, true) } else { DM_LAUNCH2(HD, BSZ, false) }

  if (D == 128 && BS == 32) { DM_LAUNCH(128, 32) }
  else if (D == 128 && BS == 16) { DM_LAUNCH(128, 16) }
  else if (D == 128 && BS == 64) { DM_LAUNCH(128, 64) }
  else if (D == 64 && BS == 32) { DM_LAUNCH(64, 32) }
  else if (D == 64 && BS == 16) { DM_LAUNCH(64, 16) }
  else if (D == 64 && BS == 64) { DM_LAUNCH(64, 64) }
#undef DM_LAUNCH
#undef DM_LAUNCH2
}
