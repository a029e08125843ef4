// Paged decode attention for gfx950 — the hot kernel of the engine.
//
// Design (MI355X-first, see cdna_hip_programming.md §B "Attention decode"):
// decode attention is HBM-bound KV streaming. One 256-thread workgroup per
// (sequence, kv_head) processes all G = Hq/Hk query heads of the group so KV is
// read exactly once. Context is processed in 128-token chunks with online
// softmax:
//   Phase A (lanes <-> tokens): K rows stream HBM->VGPR 16 B/lane; scores for
//     all G heads via fp32 dot with q staged in LDS (wave-broadcast reads);
//     chunk max -> rescale factors -> p = exp(s - m) into LDS.
//   Phase B (lanes <-> output dims): V rows stream coalesced (16 threads x
//     16 B = one 256 B row), every thread accumulates G x 8 fp32 outputs in
//     registers; partials across the 16 token-groups are tree-reduced in LDS
//     once at the end.
// Long contexts use the same kernel in partitioned (flash-decoding split-K)
// mode: grid.z partitions write (acc, m, l) to workspace and a small reduce
// kernel combines — reference analogue: paged_attention_v2 + v2_reduce
// (parallax_extensions/kernels/paged_attention/, Metal); fresh HIP design.
//
// Supports sliding window, logit softcap and attention sinks (gpt-oss).

#include "common.h"

#define ATTN_THREADS 256
#define CHUNK_TOKENS 128

template <int HEAD_DIM, int BLOCK_SIZE, int GMAX, bool PARTITIONED>
__global__ __launch_bounds__(ATTN_THREADS) void paged_attention_kernel(
    uint16_t* __restrict__ out,        // [B, Hq, D] bf16 (final mode)
    float* __restrict__ tmp_acc,       // [B, Hq, P, D] (partitioned mode)
    float* __restrict__ tmp_ml,        // [B, Hq, P, 2] (m, l)
    const uint16_t* __restrict__ q,    // [B, Hq, D]
    const uint16_t* __restrict__ k_cache,  // [NB, Hk, BS, D]
    const uint16_t* __restrict__ v_cache,
    const int* __restrict__ block_tables,  // [B, max_blocks]
    const int* __restrict__ seq_lens,      // [B]
    const int max_blocks,
    const int Hk, const int G,
    const int64_t q_stride,
    const float scale,
    const int sliding_window,           // <=0: full
    const float softcap,                // <=0: off
    const float* __restrict__ sinks,    // [Hq] or nullptr
    const int part_tokens) {            // tokens per partition (PARTITIONED)
  constexpr int HALF = HEAD_DIM / 2;       // elems per (token, half) lane
  constexpr int KVECS = HALF / 8;          // bf16x8 loads per lane
  constexpr int DC = HEAD_DIM / 8;         // dim-chunks in phase B
  constexpr int NT_PAR = ATTN_THREADS / DC;  // token-parallel groups in phase B

  const int seq = blockIdx.y;
  const int hk = blockIdx.x;
  const int L = seq_lens[seq];

  int tok_begin = 0, tok_end = L;
  if (sliding_window > 0) tok_begin = max(0, L - sliding_window);
  if (PARTITIONED) {
    const int p = blockIdx.z;
    tok_begin = max(tok_begin, p * part_tokens);
    tok_end = min(tok_end, (p + 1) * part_tokens);
  }

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;

  if (tok_begin >= tok_end) {
    // empty partition: record neutral (l=0 -> reduce kernel skips the acc read)
    if (PARTITIONED) {
      const int p = blockIdx.z;
      for (int g = tid; g < G; g += ATTN_THREADS) {
        const size_t hq = (size_t)hk * G + g;
        float* ml = tmp_ml + (((size_t)seq * Hk * G + hq) * gridDim.z + p) * 2;
        ml[0] = -1e30f;
        ml[1] = 0.f;
      }
    }
    return;
  }

  __shared__ float q_lds[GMAX][HEAD_DIM];
  __shared__ float p_lds[GMAX][CHUNK_TOKENS];
  __shared__ float redw[GMAX][4][HEAD_DIM];  // per-wave output partials
  __shared__ float m_lds[GMAX], l_lds[GMAX], rescale_lds[GMAX];
  __shared__ float wmax_lds[4][GMAX], wsum_lds[4][GMAX];

  // ---- load q (scaled) into LDS -------------------------------------------------
  for (int i = tid; i < G * HEAD_DIM; i += ATTN_THREADS) {
    const int g = i / HEAD_DIM, d = i % HEAD_DIM;
    q_lds[g][d] =
        bf16_bits_to_f32(q[(size_t)seq * q_stride + (hk * G + g) * HEAD_DIM + d]) * scale;
  }
  if (tid < GMAX) {
    m_lds[tid] = -1e30f;
    l_lds[tid] = 0.f;
  }
  __syncthreads();

  // phase-B accumulators: this thread owns dim-chunk dc for token-group tp
  const int dc = tid % DC;
  const int tp = tid / DC;
  float acc[GMAX][8];
#pragma unroll
  for (int g = 0; g < GMAX; ++g)
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[g][j] = 0.f;

  const int* btab = block_tables + (size_t)seq * max_blocks;

  const int chunk_first = tok_begin / CHUNK_TOKENS;
  const int chunk_last = (tok_end - 1) / CHUNK_TOKENS;

  for (int chunk = chunk_first; chunk <= chunk_last; ++chunk) {
    const int base_tok = chunk * CHUNK_TOKENS;

    // ---- phase A: scores ------------------------------------------------------
    const int t = tid >> 1;           // token within chunk
    const int half = tid & 1;         // which half of HEAD_DIM
    const int tok = base_tok + t;
    const bool valid = (tok >= tok_begin) && (tok < tok_end);

    bf16x8 kvec[KVECS];
    if (valid) {
      const int blk = btab[tok / BLOCK_SIZE];
      const int off = tok % BLOCK_SIZE;
      const uint16_t* krow =
          k_cache + (((size_t)blk * Hk + hk) * BLOCK_SIZE + off) * HEAD_DIM +
          half * HALF;
#pragma unroll
      for (int i = 0; i < KVECS; ++i) kvec[i] = load_bf16x8(krow + i * 8);
    }

    float sreg[GMAX];
#pragma unroll
    for (int g = 0; g < GMAX; ++g) {
      if (g >= G) break;
      float s = 0.f;
      if (valid) {
#pragma unroll
        for (int i = 0; i < KVECS; ++i)
          s += bf16x8_dot(kvec[i], &q_lds[g][half * HALF + i * 8]);
      }
      // combine the two halves (adjacent lanes)
      s += __shfl_xor(s, 1, WAVE_SIZE);
      if (softcap > 0.f) s = softcap * tanhf(s / softcap);
      if (!valid) s = -1e30f;
      sreg[g] = s;
      const float wm = wave_reduce_max(s);
      if (lane == 0) wmax_lds[wid][g] = wm;
    }
    __syncthreads();

    if (tid < G) {
      const float m_chunk = fmaxf(fmaxf(wmax_lds[0][tid], wmax_lds[1][tid]),
                                  fmaxf(wmax_lds[2][tid], wmax_lds[3][tid]));
      const float m_old = m_lds[tid];
      const float m_new = fmaxf(m_old, m_chunk);
      const float r = __expf(m_old - m_new);
      rescale_lds[tid] = r;
      l_lds[tid] *= r;
      m_lds[tid] = m_new;
    }
    __syncthreads();

#pragma unroll
    for (int g = 0; g < GMAX; ++g) {
      if (g >= G) break;
      const float p = valid ? __expf(sreg[g] - m_lds[g]) : 0.f;
      if (half == 0) p_lds[g][t] = p;
      const float ws = wave_reduce_sum(half == 0 ? p : 0.f);
      if (lane == 0) wsum_lds[wid][g] = ws;
    }
    __syncthreads();
    if (tid < G)
      l_lds[tid] += wsum_lds[0][tid] + wsum_lds[1][tid] + wsum_lds[2][tid] +
                    wsum_lds[3][tid];

    // ---- phase B: PV accumulate --------------------------------------------------
#pragma unroll
    for (int g = 0; g < GMAX; ++g) {
      if (g >= G) break;
      const float r = rescale_lds[g];
#pragma unroll
      for (int j = 0; j < 8; ++j) acc[g][j] *= r;
    }
    // preload every V vector for this chunk first so the HBM latencies
    // overlap (a guarded load inside the FMA loop serializes ~900 cy each)
    constexpr int VIT = CHUNK_TOKENS / NT_PAR;
    bf16x8 vv[VIT];
#pragma unroll
    for (int it = 0; it < VIT; ++it) {
      const int tok2 = base_tok + tp + it * NT_PAR;
      if (tok2 >= tok_begin && tok2 < tok_end) {
        // v cache is transposed [NB, Hk, D, BS]: this token's 8 d-elements
        // are strided by BLOCK_SIZE
        const int blk = btab[tok2 / BLOCK_SIZE];
        const int off = tok2 % BLOCK_SIZE;
        const size_t vbase =
            (((size_t)blk * Hk + hk) * HEAD_DIM + dc * 8) * BLOCK_SIZE + off;
        alignas(16) uint16_t e[8];
#pragma unroll
        for (int j = 0; j < 8; ++j)
          e[j] = v_cache[vbase + (size_t)j * BLOCK_SIZE];
        vv[it].raw = *reinterpret_cast<const int4*>(e);
      } else {
        vv[it].raw = make_int4(0, 0, 0, 0);
      }
    }
#pragma unroll
    for (int it = 0; it < VIT; ++it) {
      const int tt = tp + it * NT_PAR;
      float vf[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) vf[j] = bf16x8_get(vv[it], j);
#pragma unroll
      for (int g = 0; g < GMAX; ++g) {
        if (g >= G) break;
        const float p = p_lds[g][tt];
#pragma unroll
        for (int j = 0; j < 8; ++j) acc[g][j] = fmaf(p, vf[j], acc[g][j]);
      }
    }
    __syncthreads();  // p_lds reused next chunk
  }

  // ---- final reduce: shfl across the tp groups inside each wave (no LDS),
  // then one barrier to combine the 4 wave partials ------------------------------
#pragma unroll
  for (int g = 0; g < GMAX; ++g) {
    if (g >= G) break;
    for (int off = DC; off < WAVE_SIZE; off <<= 1) {
#pragma unroll
      for (int j = 0; j < 8; ++j)
        acc[g][j] += __shfl_xor(acc[g][j], off, WAVE_SIZE);
    }
    if (lane < DC) {
#pragma unroll
      for (int j = 0; j < 8; ++j) redw[g][wid][dc * 8 + j] = acc[g][j];
    }
  }
  __syncthreads();

  // one thread per (g, dc) combines the 4 wave partials and writes out
  const int g2 = tid / DC;
  const int dc2 = tid % DC;
  if (g2 < G) {
    const size_t hq = (size_t)hk * G + g2;
    float vals[8];
#pragma unroll
    for (int j = 0; j < 8; ++j)
      vals[j] = redw[g2][0][dc2 * 8 + j] + redw[g2][1][dc2 * 8 + j] +
                redw[g2][2][dc2 * 8 + j] + redw[g2][3][dc2 * 8 + j];
    if (PARTITIONED) {
      const int p = blockIdx.z;
      float* dst =
          tmp_acc + (((size_t)seq * Hk * G + hq) * gridDim.z + p) * HEAD_DIM;
#pragma unroll
      for (int j = 0; j < 8; ++j) dst[dc2 * 8 + j] = vals[j];
      if (dc2 == 0) {
        float* ml = tmp_ml + (((size_t)seq * Hk * G + hq) * gridDim.z + p) * 2;
        ml[0] = m_lds[g2];
        ml[1] = l_lds[g2];
      }
    } else {
      float l = l_lds[g2];
      if (sinks != nullptr) l += __expf(sinks[hq] - m_lds[g2]);
      const float inv = 1.f / l;
#pragma unroll
      for (int j = 0; j < 8; ++j) vals[j] *= inv;
      store_bf16x8(out + ((size_t)seq * Hk * G + hq) * HEAD_DIM + dc2 * 8, vals);
    }
  }
}

// combine partitions: out[b,h,:] = sum_p acc[p] * exp(m_p - M) / L
template <int HEAD_DIM>
__global__ void paged_attention_reduce_kernel(
    uint16_t* __restrict__ out,       // [B, Hq, D]
    const float* __restrict__ tmp_acc,  // [B, Hq, P, D]
    const float* __restrict__ tmp_ml,   // [B, Hq, P, 2]
    const int num_parts,
    const float* __restrict__ sinks,    // [Hq] or nullptr
    const int Hq) {
  const int seq = blockIdx.y;
  const int h = blockIdx.x;
  const size_t base = (size_t)seq * Hq + h;
  __shared__ float m_shared, l_shared;
  // thread 0 computes global max + denominator (num_parts is small)
  if (threadIdx.x == 0) {
    float M = -1e30f;
    for (int p = 0; p < num_parts; ++p)
      M = fmaxf(M, tmp_ml[(base * num_parts + p) * 2]);
    float L = 0.f;
    for (int p = 0; p < num_parts; ++p) {
      const float mp = tmp_ml[(base * num_parts + p) * 2];
      const float lp = tmp_ml[(base * num_parts + p) * 2 + 1];
      L += lp * __expf(mp - M);
    }
    if (sinks != nullptr) L += __expf(sinks[h] - M);
    m_shared = M;
    l_shared = L;
  }
  __syncthreads();
  const float M = m_shared;
  const float inv = 1.f / l_shared;
  for (int d0 = threadIdx.x * 8; d0 < HEAD_DIM; d0 += blockDim.x * 8) {
    float vals[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    for (int p = 0; p < num_parts; ++p) {
      if (tmp_ml[(base * num_parts + p) * 2 + 1] <= 0.f) continue;  // empty part
      const float w = __expf(tmp_ml[(base * num_parts + p) * 2] - M);
      const float* src = tmp_acc + (base * num_parts + p) * HEAD_DIM + d0;
#pragma unroll
      for (int j = 0; j < 8; ++j) vals[j] = fmaf(w, src[j], vals[j]);
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) vals[j] *= inv;
    store_bf16x8(out + base * HEAD_DIM + d0, vals);
  }
}

// ---- host-side dispatch ------------------------------------------------------------

#define DISPATCH_G(GVAL, ...)                       \
  if (G <= 1) {                                     \
    constexpr int GMAX = 1;                         \
    __VA_ARGS__;                                    \
  } else if (G <= 2) {                              \
    constexpr int GMAX = 2;                         \
    __VA_ARGS__;                                    \
  } else if (G <= 4) {                              \
    constexpr int GMAX = 4;                         \
    __VA_ARGS__;                                    \
  } else if (G <= 8) {                              \
    constexpr int GMAX = 8;                         \
    __VA_ARGS__;                                    \
  } else if (G <= 16) {                             \
    constexpr int GMAX = 16;                        \
    __VA_ARGS__;                                    \
  }

#define DISPATCH_HEAD_DIM(D, ...)      \
  if (D == 128) {                      \
    constexpr int HEAD_DIM = 128;      \
    __VA_ARGS__;                       \
  } else if (D == 64) {                \
    constexpr int HEAD_DIM = 64;       \
    __VA_ARGS__;                       \
  }

#define DISPATCH_BS(BSV, ...)          \
  if (BSV == 32) {                     \
    constexpr int BLOCK_SIZE = 32;     \
    __VA_ARGS__;                       \
  } else if (BSV == 16) {              \
    constexpr int BLOCK_SIZE = 16;     \
    __VA_ARGS__;                       \
  } else if (BSV == 64) {              \
    constexpr int BLOCK_SIZE = 64;     \
    __VA_ARGS__;                       \
  }

extern "C" void launch_paged_attention_reduce(
    void* out, const float* tmp_acc, const float* tmp_ml, int B, int Hq, int D,
    int num_parts, const float* sinks, hipStream_t stream) {
  dim3 rgrid(Hq, B, 1);
  if (D == 128)
    paged_attention_reduce_kernel<128><<<rgrid, 64, 0, stream>>>(
        (uint16_t*)out, tmp_acc, tmp_ml, num_parts, sinks, Hq);
  else if (D == 64)
    paged_attention_reduce_kernel<64><<<rgrid, 64, 0, stream>>>(
        (uint16_t*)out, tmp_acc, tmp_ml, num_parts, sinks, Hq);
  else if (D == 512)  // MLA latent dim
    paged_attention_reduce_kernel<512><<<rgrid, 64, 0, stream>>>(
        (uint16_t*)out, tmp_acc, tmp_ml, num_parts, sinks, Hq);
}

extern "C" void launch_paged_attention_decode(
    void* out, const void* q, const void* k_cache, const void* v_cache,
    const int* block_tables, const int* seq_lens, int B, int Hq, int Hk, int D,
    int BS, int max_blocks, int64_t q_stride, float scale, int sliding_window,
    float softcap, const float* sinks, int num_parts, int part_tokens,
    float* tmp_acc, float* tmp_ml, hipStream_t stream, bool* launched) {
  const int G = Hq / Hk;
  *launched = false;
  if (num_parts <= 1) {
    dim3 grid(Hk, B, 1);
    DISPATCH_HEAD_DIM(D, DISPATCH_BS(BS, DISPATCH_G(G, {
      paged_attention_kernel<HEAD_DIM, BLOCK_SIZE, GMAX, false>
          <<<grid, ATTN_THREADS, 0, stream>>>(
              (uint16_t*)out, nullptr, nullptr, (const uint16_t*)q,
              (const uint16_t*)k_cache, (const uint16_t*)v_cache, block_tables,
              seq_lens, max_blocks, Hk, G, q_stride, scale, sliding_window,
              softcap, sinks, 0);
      *launched = true;
    })));
  } else {
    dim3 grid(Hk, B, num_parts);
    DISPATCH_HEAD_DIM(D, DISPATCH_BS(BS, DISPATCH_G(G, {
      paged_attention_kernel<HEAD_DIM, BLOCK_SIZE, GMAX, true>
          <<<grid, ATTN_THREADS, 0, stream>>>(
              nullptr, tmp_acc, tmp_ml, (const uint16_t*)q,
              (const uint16_t*)k_cache, (const uint16_t*)v_cache, block_tables,
              seq_lens, max_blocks, Hk, G, q_stride, scale, sliding_window,
              softcap, sinks, part_tokens);
      *launched = true;
    })));
  }
}
