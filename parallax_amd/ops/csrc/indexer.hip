// Sparse-attention indexer kernels (DSA / MSA) for gfx950.
//
// Reference analogues (Metal): kernels/dsa/dsa_indexer.metal (176 LoC) and
// kernels/msa/msa_indexer.metal (232 LoC) — score passes that pick which
// context tokens the sparse decode kernels attend to. At 256k context the
// score pass streams the whole index/K cache, so these are written as real
// CDNA4 kernels (MFMA for the DSA q.k scores, coalesced streaming reduction
// for the MSA block pool) instead of the round-1 torch compositions.
//
//   dsa_indexer_scores: score[t] = sum_h w[h] * relu(q[h] . k[t])  (shared
//     keys across heads, DeepSeek-V3.2 geometry: Hi<=64, Di=128). One
//     workgroup per (seq, 64-token tile); S = Q.K^T on mfma_16x16x32, the
//     relu+head-weight reduction folds inside the accumulator registers.
//   store_indexer_cache: slot scatter of new index keys (graph-safe: pad
//     slots -1 land in the cache's trash block, chosen by the host wrapper).
//   msa_block_scores: mean-pooled K per sparse block dotted with the
//     head-mean query (MiniMax-M3 phase 1).
//   msa_topk_tokens: per-seq top-k block selection expanded to sorted token
//     positions with always-kept init/local blocks (phase 2; one workgroup
//     per sequence, selection over <=4096 sparse blocks in LDS).

#include "common.h"

#define IDX_THREADS 256
#define IDX_KTILE 64

// ---- DSA: weighted relu(q.k) scores over the paged index cache ----------------
// q_index [B, Hi, Di] bf16; index_cache [NB, BS, Di] bf16 (keys shared across
// heads); head_weights [B, Hi] fp32; out scores [B, max_ctx] fp32 (-inf pad).

template <int DI>
__global__ __launch_bounds__(IDX_THREADS) void dsa_indexer_scores_kernel(
    float* __restrict__ scores,           // [B, max_ctx]
    const uint16_t* __restrict__ q_index, // [B, Hi, DI]
    const uint16_t* __restrict__ cache,   // [NB, BS, DI]
    const float* __restrict__ head_w,     // [B, Hi]
    const int* __restrict__ block_tables, // [B, max_blocks]
    const int* __restrict__ seq_lens,
    const int Hi, const int BS, const int max_blocks, const int max_ctx) {
  const int seq = blockIdx.y;
  const int t0 = blockIdx.x * IDX_KTILE;
  const int L = seq_lens[seq];
  if (t0 >= max_ctx) return;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int l15 = lane & 15;
  const int l4 = lane >> 4;

  if (t0 >= L) {  // pad region of the batch: -inf rows
    for (int t = t0 + tid; t < min(t0 + IDX_KTILE, max_ctx); t += IDX_THREADS)
      scores[(size_t)seq * max_ctx + t] = -3.0e30f;
    return;
  }

  __shared__ uint16_t Ql[64 * DI];        // head rows, swizzled
  __shared__ uint16_t Kl[IDX_KTILE * DI]; // token rows, swizzled
  __shared__ float red[4][IDX_KTILE];

  // stage q heads (<= 64 rows; zero-pad)
  {
    const int qrow = tid & 63;
    const int dv = tid >> 6;  // 4 chunks of DI/4
#pragma unroll
    for (int c = 0; c < DI / 32; ++c) {
      const int d = dv * (DI / 4) + c * 8;
      int4 val = make_int4(0, 0, 0, 0);
      if (qrow < Hi)
        val = *reinterpret_cast<const int4*>(
            q_index + ((size_t)seq * Hi + qrow) * DI + d);
      const int byte = swz(qrow * DI * 2 + d * 2, qrow);
      *reinterpret_cast<int4*>(reinterpret_cast<char*>(Ql) + byte) = val;
    }
  }
  // stage the K tile (paged lookup per token row)
  {
    const int trow = tid & 63;
    const int dv = tid >> 6;
    const int gtok = t0 + trow;
    size_t row_off = 0;
    const bool ok = gtok < L;
    if (ok) {
      const int blk = block_tables[(size_t)seq * max_blocks + gtok / BS];
      row_off = ((size_t)blk * BS + gtok % BS) * DI;
    }
#pragma unroll
    for (int c = 0; c < DI / 32; ++c) {
      const int d = dv * (DI / 4) + c * 8;
      int4 val = make_int4(0, 0, 0, 0);
      if (ok) val = *reinterpret_cast<const int4*>(cache + row_off + d);
      const int byte = swz(trow * DI * 2 + d * 2, trow);
      *reinterpret_cast<int4*>(reinterpret_cast<char*>(Kl) + byte) = val;
    }
  }
  __syncthreads();

  // wave w scores head rows [16w, 16w+16) x the 64-token tile:
  // S[h][t] on four 16x16 MFMA n-tiles; head weight + relu fold here.
  float part[4];  // per-lane partial: 4 token cols x (4 head rows summed)
  const int hrow0 = wid * 16;
#pragma unroll
  for (int nt = 0; nt < 4; ++nt) {
    f32x4v acc = {};
#pragma unroll
    for (int s = 0; s < DI / 32; ++s) {
      const int hrow = hrow0 + l15;
      const int qa = swz(hrow * DI * 2 + s * 64 + l4 * 16, hrow);
      const bf16x8v afrag = *reinterpret_cast<const bf16x8v*>(
          reinterpret_cast<const char*>(Ql) + qa);
      const int trow = nt * 16 + l15;
      const int kb = swz(trow * DI * 2 + s * 64 + l4 * 16, trow);
      const bf16x8v bfrag = *reinterpret_cast<const bf16x8v*>(
          reinterpret_cast<const char*>(Kl) + kb);
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc, 0, 0, 0);
    }
    // D layout: row m (head) = l4*4 + r, col n (token) = l15
    float p = 0.f;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int h = hrow0 + l4 * 4 + r;
      const float s = fmaxf(acc[r], 0.f);  // relu
      p += (h < Hi) ? s * head_w[(size_t)seq * Hi + h] : 0.f;
    }
    part[nt] = p;
  }
  // reduce the 4 head-quadrant lanes (l4) per token column, then across waves
#pragma unroll
  for (int nt = 0; nt < 4; ++nt) {
    float p = part[nt];
    p += __shfl_xor(p, 16, WAVE_SIZE);
    p += __shfl_xor(p, 32, WAVE_SIZE);
    if (l4 == 0) red[wid][nt * 16 + l15] = p;
  }
  __syncthreads();
  for (int t = tid; t < IDX_KTILE; t += IDX_THREADS) {
    const int gtok = t0 + t;
    if (gtok >= max_ctx) break;
    const float v = red[0][t] + red[1][t] + red[2][t] + red[3][t];
    scores[(size_t)seq * max_ctx + gtok] = (gtok < L) ? v : -3.0e30f;
  }
}

// ---- store_indexer_cache: slot scatter ----------------------------------------
// keys [T, DI] bf16 -> cache [NB, BS, DI]; slot < 0 redirects to trash_slot.

__global__ void store_indexer_cache_kernel(
    uint16_t* __restrict__ cache, const uint16_t* __restrict__ keys,
    const int64_t* __restrict__ slot_mapping, const int64_t trash_slot,
    const int DI) {
  const int t = blockIdx.x;
  int64_t slot = slot_mapping[t];
  if (slot < 0) slot = trash_slot;
  const uint16_t* src = keys + (size_t)t * DI;
  uint16_t* dst = cache + (size_t)slot * DI;
  for (int d = threadIdx.x * 8; d < DI; d += blockDim.x * 8)
    *reinterpret_cast<int4*>(dst + d) =
        *reinterpret_cast<const int4*>(src + d);
}

// ---- MSA phase 1: mean-pooled block scores ------------------------------------
// score[b][sb] = mean_{t in sb, h} K[t,h,:] . mean_h' q[h',:]
// grid (max_sparse_blocks, B); 256 threads stream the sparse block's K rows.

template <int D>
__global__ __launch_bounds__(IDX_THREADS) void msa_block_scores_kernel(
    float* __restrict__ out,             // [B, max_sparse_blocks]
    const uint16_t* __restrict__ q,      // [B, Hq, D]
    const uint16_t* __restrict__ k_cache,// [NB, Hk, BS, D]
    const int* __restrict__ block_tables,
    const int* __restrict__ seq_lens,
    const int Hq, const int Hk, const int BS, const int max_blocks,
    const int sparse_block, const int max_sparse_blocks) {
  const int seq = blockIdx.y;
  const int sb = blockIdx.x;
  const int L = seq_lens[seq];
  const int nsb = (L + sparse_block - 1) / sparse_block;
  if (sb >= nsb) {
    if (threadIdx.x == 0 && sb < max_sparse_blocks)
      out[(size_t)seq * max_sparse_blocks + sb] = -3.0e30f;
    return;
  }
  const int tid = threadIdx.x;
  const int d = tid & (D - 1);       // D is a power of two (64/128)
  const int rows_per_pass = IDX_THREADS / D;
  const int row_in_pass = tid / D;

  // head-mean query component for this thread's d
  float qm = 0.f;
  for (int h = 0; h < Hq; ++h)
    qm += bf16_bits_to_f32(q[((size_t)seq * Hq + h) * D + d]);
  qm /= Hq;

  const int tok0 = sb * sparse_block;
  const int tok_end = min(tok0 + sparse_block, L);
  float acc = 0.f;  // sum over (token, kv head) of K[t,h,d]
  for (int t = tok0 + row_in_pass; t < tok_end; t += rows_per_pass) {
    const int blk = block_tables[(size_t)seq * max_blocks + t / BS];
    const size_t base = (((size_t)blk * Hk) * BS + t % BS) * D;
    for (int h = 0; h < Hk; ++h)
      acc += bf16_bits_to_f32(k_cache[base + (size_t)h * BS * D + d]);
  }
  // mean over tokens*heads, dot with qm, sum over d
  const float n = (float)(tok_end - tok0) * Hk;
  float partial = (acc / n) * qm;
  __shared__ float lds[8];
  const int lane = tid & 63, wid = tid >> 6;
  partial = wave_reduce_sum(partial);
  if (lane == 0) lds[wid] = partial;
  __syncthreads();
  if (tid == 0) {
    // threads sharing a d hold disjoint token subsets, so the plain sum is
    // already the full mean-pool dot product
    float s = 0.f;
    for (int w = 0; w < IDX_THREADS / 64; ++w) s += lds[w];
    out[(size_t)seq * max_sparse_blocks + sb] = s;
  }
}

// ---- MSA phase 2: top-k blocks -> sorted token positions ----------------------
// One workgroup per sequence. keep[] bitmap in LDS; selection is an O(k*nsb)
// argmax sweep (k and nsb are small: <=64 and <=4096).

#define MSA_MAX_SB 4096

__global__ __launch_bounds__(IDX_THREADS) void msa_topk_tokens_kernel(
    int64_t* __restrict__ out,           // [B, max_positions] (-1 pad)
    const float* __restrict__ scores,    // [B, max_sparse_blocks]
    const int* __restrict__ seq_lens,
    const int sparse_block, const int max_sparse_blocks,
    const int topk_blocks, const int init_blocks, const int local_blocks,
    const int max_positions) {
  const int seq = blockIdx.x;
  const int L = seq_lens[seq];
  const int nsb = min((L + sparse_block - 1) / sparse_block, MSA_MAX_SB);
  const int tid = threadIdx.x;

  __shared__ uint8_t keep[MSA_MAX_SB];
  __shared__ int kept_off[MSA_MAX_SB + 1];
  for (int b = tid; b < nsb; b += IDX_THREADS) {
    const bool always = (b < init_blocks) || (b >= nsb - local_blocks);
    keep[b] = always ? 1 : 0;
  }
  __syncthreads();

  // k parallel argmax passes over the non-kept blocks (k, nsb small; each
  // pass is a strided scan + wave/block argmax reduction)
  __shared__ float wv[4];
  __shared__ int wi[4];
  const int lane = tid & 63, wid = tid >> 6;
  for (int it = 0; it < topk_blocks; ++it) {
    float v = -3.0e30f;
    int vi = -1;
    for (int b = tid; b < nsb; b += IDX_THREADS) {
      if (keep[b]) continue;
      const float s = scores[(size_t)seq * max_sparse_blocks + b];
      if (s > v) {
        v = s;
        vi = b;
      }
    }
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      const float ov = __shfl_xor(v, off, WAVE_SIZE);
      const int oi = __shfl_xor(vi, off, WAVE_SIZE);
      if (ov > v || (ov == v && oi >= 0 && oi < vi)) {
        v = ov;
        vi = oi;
      }
    }
    if (lane == 0) {
      wv[wid] = v;
      wi[wid] = vi;
    }
    __syncthreads();
    if (tid == 0) {
      float bv = -3.0e30f;
      int bi = -1;
      for (int w = 0; w < 4; ++w)
        if (wi[w] >= 0 && (wv[w] > bv || (wv[w] == bv && wi[w] < bi))) {
          bv = wv[w];
          bi = wi[w];
        }
      if (bi >= 0) keep[bi] = 1;
      wi[0] = bi;  // broadcast the stop signal
    }
    __syncthreads();
    if (wi[0] < 0) break;
    __syncthreads();
  }

  // exclusive prefix over kept block token counts -> write offsets
  if (tid == 0) {
    int off = 0;
    for (int b = 0; b < nsb; ++b) {
      kept_off[b] = off;
      if (keep[b]) off += min(sparse_block, L - b * sparse_block);
    }
    kept_off[nsb] = off;
  }
  __syncthreads();
  const int total = kept_off[nsb];
  for (int b = 0; b < nsb; ++b) {
    if (!keep[b]) continue;
    const int t0 = b * sparse_block;
    const int cnt = min(sparse_block, L - t0);
    for (int j = tid; j < cnt; j += IDX_THREADS)
      out[(size_t)seq * max_positions + kept_off[b] + j] = t0 + j;
  }
  for (int j = total + tid; j < max_positions; j += IDX_THREADS)
    out[(size_t)seq * max_positions + j] = -1;
}

// ---- launchers ----------------------------------------------------------------

extern "C" void launch_dsa_indexer_scores(
    float* scores, const void* q_index, const void* cache, const float* head_w,
    const int* block_tables, const int* seq_lens, int B, int Hi, int Di,
    int BS, int max_blocks, int max_ctx, hipStream_t stream, bool* launched) {
  *launched = false;
  if (Hi > 64) return;
  dim3 grid(ceil_div(max_ctx, IDX_KTILE), B, 1);
  if (Di == 128) {
    dsa_indexer_scores_kernel<128><<<grid, IDX_THREADS, 0, stream>>>(
        scores, (const uint16_t*)q_index, (const uint16_t*)cache, head_w,
        block_tables, seq_lens, Hi, BS, max_blocks, max_ctx);
    *launched = true;
  } else if (Di == 64) {
    dsa_indexer_scores_kernel<64><<<grid, IDX_THREADS, 0, stream>>>(
        scores, (const uint16_t*)q_index, (const uint16_t*)cache, head_w,
        block_tables, seq_lens, Hi, BS, max_blocks, max_ctx);
    *launched = true;
  }
}

extern "C" void launch_store_indexer_cache(
    void* cache, const void* keys, const int64_t* slot_mapping, int T,
    int64_t trash_slot, int DI, hipStream_t stream) {
  if (T <= 0) return;
  store_indexer_cache_kernel<<<T, 64, 0, stream>>>(
      (uint16_t*)cache, (const uint16_t*)keys, slot_mapping, trash_slot, DI);
}

extern "C" void launch_msa_block_scores(
    float* out, const void* q, const void* k_cache, const int* block_tables,
    const int* seq_lens, int B, int Hq, int Hk, int D, int BS, int max_blocks,
    int sparse_block, int max_sparse_blocks, hipStream_t stream,
    bool* launched) {
  *launched = false;
  dim3 grid(max_sparse_blocks, B, 1);
  if (D == 128) {
    msa_block_scores_kernel<128><<<grid, IDX_THREADS, 0, stream>>>(
        out, (const uint16_t*)q, (const uint16_t*)k_cache, block_tables,
        seq_lens, Hq, Hk, BS, max_blocks, sparse_block, max_sparse_blocks);
    *launched = true;
  } else if (D == 64) {
    msa_block_scores_kernel<64><<<grid, IDX_THREADS, 0, stream>>>(
        out, (const uint16_t*)q, (const uint16_t*)k_cache, block_tables,
        seq_lens, Hq, Hk, BS, max_blocks, sparse_block, max_sparse_blocks);
    *launched = true;
  }
}

extern "C" void launch_msa_topk_tokens(
    int64_t* out, const float* scores, const int* seq_lens, int B,
    int sparse_block, int max_sparse_blocks, int topk_blocks, int init_blocks,
    int local_blocks, int max_positions, hipStream_t stream) {
  msa_topk_tokens_kernel<<<B, IDX_THREADS, 0, stream>>>(
      out, scores, seq_lens, sparse_block, max_sparse_blocks, topk_blocks,
      init_blocks, local_blocks, max_positions);
}
