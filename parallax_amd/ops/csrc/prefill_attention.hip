// Flash-style varlen prefill attention over the paged KV cache — MFMA/CDNA4.
//
// Geometry: one 256-thread workgroup (4 waves) per (request, 32-row q-tile,
// query head). KV walks in 64-token tiles staged in LDS (XOR-swizzled rows for
// conflict-free ds_read_b128 fragment loads — cdna_hip_programming.md §6 G4):
//   S^T[64k x 32q] = K_tile · Q^T     (mfma_f32_16x16x32_bf16; A = K rows
//                                      natural, B = Q rows natural — swapped
//                                      operands avoid any transpose for QK^T)
//   online softmax per q column (acc layout col=lane&15 = q, row = k)
//   O^T[128d x 32q] += V^T · P        (V staged TRANSPOSED during global->LDS;
//                                      P written [q][k] so both A and B
//                                      fragments are contiguous ds_read_b128)
// Causal + prefix (seq_len > query_len), sliding window, softcap, sinks.
//
// Reference analogue: the reference has no prefill kernel (it delegates prefill
// to SGLang/MLX); this is the MI355X-native equivalent demanded by the
// continuous-batching engine. Fresh design per the CDNA4 attention recipe.

#include "common.h"

#define PF_THREADS 256
#define QTILE 32
#define KTILE 64

template <int HEAD_DIM, bool KV_FP8>
__global__ __launch_bounds__(PF_THREADS) void prefill_attention_kernel(
    uint16_t* __restrict__ out,            // [T, Hq, D]
    const uint16_t* __restrict__ q,        // [T, Hq, D]
    const void* __restrict__ k_cache_v,    // [NB, Hk, BS, D] bf16 | fp8
    const void* __restrict__ v_cache_v,
    const int* __restrict__ block_tables,  // [B, max_blocks]
    const int* __restrict__ seq_lens,      // [B] total ctx (prefix + new)
    const int* __restrict__ cu_q,          // [B+1] query offsets
    const int* __restrict__ tile_req,      // [n_tiles] request of each q tile
    const int* __restrict__ tile_row0,     // [n_tiles] first q row of the tile
    const int max_blocks, const int Hq, const int Hk, const int BS,
    const int64_t q_stride, const float scale, const int sliding_window,
    const float softcap, const float* __restrict__ sinks,
    const float k_scale, const float v_scale) {
  const int h = blockIdx.x;
  const int tile = blockIdx.y;
  const int req = tile_req[tile];
  const int r0 = tile_row0[tile];
  const int hk = h / (Hq / Hk);
  const int L = seq_lens[req];
  const int q0 = cu_q[req];
  const int QL = cu_q[req + 1] - q0;
  const int prefix = L - QL;
  const int* btab = block_tables + (size_t)req * max_blocks;

  __shared__ uint16_t Kl[KTILE * HEAD_DIM];      // row = token (256 B, swz)
  __shared__ uint16_t VTl[HEAD_DIM * KTILE];     // row = dim   (128 B, swz)
  __shared__ uint16_t Ql[QTILE * HEAD_DIM];      // row = q     (256 B, swz)
  __shared__ uint16_t Pl[QTILE * KTILE];         // row = q     (128 B, swz)
  __shared__ float m_s[QTILE], l_s[QTILE], resc[QTILE];
  __shared__ float wred[4][QTILE];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int l15 = lane & 15;
  const int l4 = lane >> 4;

  // ---- stage Q tile (scaled later in fp32; bf16 rows as-is) --------------------
  {
    const int qrow = tid & 31;
    const int dv = tid >> 5;  // 8 chunks of 16 elems
    const int d = dv * (HEAD_DIM / 8);
    const bool ok = r0 + qrow < QL;
#pragma unroll
    for (int c = 0; c < HEAD_DIM / 64; ++c) {  // 16 elems per chunk (2 int4)
      int4 val = make_int4(0, 0, 0, 0);
      if (ok)
        val = *reinterpret_cast<const int4*>(
            q + (size_t)(q0 + r0 + qrow) * q_stride + h * HEAD_DIM + d + c * 8);
      const int byte = swz(qrow * HEAD_DIM * 2 + (d + c * 8) * 2, qrow);
      *reinterpret_cast<int4*>(reinterpret_cast<char*>(Ql) + byte) = val;
    }
  }
  if (tid < QTILE) {
    m_s[tid] = -3.0e4f;
    l_s[tid] = 0.f;
  }

  // PV accumulators: the 4 waves split the D rows of O^T evenly
  constexpr int MT = HEAD_DIM / 16 / 4;  // M-tiles per wave (2 @D=128, 1 @D=64)
  f32x4v acc_o[MT][2] = {};

  const int qpos0 = prefix + r0;
  int kt_begin = 0;
  if (sliding_window > 0) kt_begin = max(0, qpos0 - sliding_window + 1) / KTILE;
  const int kv_limit = min(L, qpos0 + QTILE);
  const int kt_end = (kv_limit + KTILE - 1) / KTILE;

  // register double-buffered staging: next tile's loads fly during this
  // tile's QK/softmax/PV (guide §6 G15 async-STAGE split).
  // K: per-token rows (natural layout). V: the cache is TRANSPOSED
  // ([NB, Hk, D, BS]), so V^T rows stage as vectorized row-major copies —
  // no per-element transpose scatter.
  const int stg_tok = tid & 63;
  const int stg_dv = tid >> 6;
  const int stg_d = stg_dv * (HEAD_DIM / 4);
  int4 kreg[HEAD_DIM / 32];
  // V^T: threads_per_row threads cover each d row's KTILE tokens
  constexpr int VT_TPR = PF_THREADS / HEAD_DIM;        // 2 @128, 4 @64
  constexpr int VT_TOK = KTILE / VT_TPR;               // tokens per thread
  const int vt_d = tid / VT_TPR;                       // d row
  const int vt_t0 = (tid % VT_TPR) * VT_TOK;           // first token
  int4 vreg[VT_TOK / 8];

  auto load_tile = [&](int kt) {
    {
      const int gtok = kt * KTILE + stg_tok;
      const bool ok = gtok < L;
      size_t row_off = 0;
      if (ok) {
        const int blk = btab[gtok / BS];
        const int off = gtok % BS;
        row_off = (((size_t)blk * Hk + hk) * BS + off) * HEAD_DIM;
      }
#pragma unroll
      for (int c = 0; c < HEAD_DIM / 32; ++c) {  // 8 elems per step
        int4 kval = make_int4(0, 0, 0, 0);
        if (ok) {
          if (KV_FP8) {
            const uint64_t kraw = *reinterpret_cast<const uint64_t*>(
                (const uint8_t*)k_cache_v + row_off + stg_d + c * 8);
            fp8x8_to_bf16x8(kraw, k_scale,
                            reinterpret_cast<uint16_t*>(&kval));
          } else {
            kval = *reinterpret_cast<const int4*>(
                (const uint16_t*)k_cache_v + row_off + stg_d + c * 8);
          }
        }
        kreg[c] = kval;
      }
    }
#pragma unroll
    for (int c = 0; c < VT_TOK / 8; ++c) {  // 8 tokens per step (one block)
      const int gtok = kt * KTILE + vt_t0 + c * 8;
      int4 vval = make_int4(0, 0, 0, 0);
      if (gtok < L) {
        const int blk = btab[gtok / BS];
        const size_t row = (((size_t)blk * Hk + hk) * HEAD_DIM + vt_d) * BS +
                           gtok % BS;
        if (KV_FP8) {
          const uint64_t vraw = *reinterpret_cast<const uint64_t*>(
              (const uint8_t*)v_cache_v + row);
          fp8x8_to_bf16x8(vraw, v_scale,
                          reinterpret_cast<uint16_t*>(&vval));
        } else {
          vval = *reinterpret_cast<const int4*>(
              (const uint16_t*)v_cache_v + row);
        }
      }
      vreg[c] = vval;
    }
  };

  if (kt_begin < kt_end) load_tile(kt_begin);

  for (int kt = kt_begin; kt < kt_end; ++kt) {
    const int kbase = kt * KTILE;
    __syncthreads();  // previous PV finished reading VTl/Pl

    // ---- write the prefetched tile: K token rows, V^T d rows (both swz) --------
#pragma unroll
    for (int c = 0; c < HEAD_DIM / 32; ++c) {
      const int d = stg_d + c * 8;
      const int kb = swz(stg_tok * HEAD_DIM * 2 + d * 2, stg_tok);
      *reinterpret_cast<int4*>(reinterpret_cast<char*>(Kl) + kb) = kreg[c];
    }
#pragma unroll
    for (int c = 0; c < VT_TOK / 8; ++c) {
      const int vb = swz(vt_d * KTILE * 2 + (vt_t0 + c * 8) * 2, vt_d);
      *reinterpret_cast<int4*>(reinterpret_cast<char*>(VTl) + vb) = vreg[c];
    }
    __syncthreads();
    if (kt + 1 < kt_end) load_tile(kt + 1);

    // ---- S^T = K . Q^T ---------------------------------------------------------
    // wave w covers k rows [16w, 16w+16); acc col = q, row = k
    f32x4v acc_s[2] = {};
#pragma unroll
    for (int s = 0; s < HEAD_DIM / 32; ++s) {
      const int krow_i = 16 * wid + l15;
      const int ka = swz(krow_i * HEAD_DIM * 2 + s * 64 + l4 * 16, krow_i);
      const bf16x8v afrag = *reinterpret_cast<const bf16x8v*>(
          reinterpret_cast<const char*>(Kl) + ka);
#pragma unroll
      for (int n = 0; n < 2; ++n) {
        const int qrow_i = n * 16 + l15;
        const int qb = swz(qrow_i * HEAD_DIM * 2 + s * 64 + l4 * 16, qrow_i);
        const bf16x8v bfrag = *reinterpret_cast<const bf16x8v*>(
            reinterpret_cast<const char*>(Ql) + qb);
        acc_s[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag,
                                                           acc_s[n], 0, 0, 0);
      }
    }

    // ---- scale, softcap, causal/window mask; per-q-column max ------------------
    float mx[2] = {-3.0e4f, -3.0e4f};
#pragma unroll
    for (int n = 0; n < 2; ++n) {
      const int qcol = n * 16 + l15;
      const int qpos = qpos0 + qcol;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int kglob = kbase + 16 * wid + l4 * 4 + r;
        float s = acc_s[n][r] * scale;
        if (softcap > 0.f) s = softcap * tanhf(s / softcap);
        const bool visible =
            (kglob <= qpos) && (kglob < L) && (r0 + qcol < QL) &&
            (sliding_window <= 0 || kglob > qpos - sliding_window);
        s = visible ? s : -3.0e4f;
        acc_s[n][r] = s;
        mx[n] = fmaxf(mx[n], s);
      }
      // lanes {q, q+16, q+32, q+48} hold the same q column
      mx[n] = fmaxf(mx[n], __shfl_xor(mx[n], 16, WAVE_SIZE));
      mx[n] = fmaxf(mx[n], __shfl_xor(mx[n], 32, WAVE_SIZE));
    }
    if (lane < 32) wred[wid][lane] = (lane < 16) ? mx[0] : mx[1];
    __syncthreads();
    if (tid < QTILE) {
      const float m_chunk = fmaxf(fmaxf(wred[0][tid], wred[1][tid]),
                                  fmaxf(wred[2][tid], wred[3][tid]));
      const float m_new = fmaxf(m_s[tid], m_chunk);
      resc[tid] = __expf(m_s[tid] - m_new);
      m_s[tid] = m_new;
    }
    __syncthreads();

    // ---- p = exp(s - m); write P[q][k]; per-q-column sum -----------------------
    float sm[2] = {0.f, 0.f};
#pragma unroll
    for (int n = 0; n < 2; ++n) {
      const int qcol = n * 16 + l15;
      const float m = m_s[qcol];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const float p = __expf(acc_s[n][r] - m);
        sm[n] += p;
        const int kk = 16 * wid + l4 * 4 + r;
        const int pb = swz(qcol * KTILE * 2 + kk * 2, qcol);
        *reinterpret_cast<uint16_t*>(reinterpret_cast<char*>(Pl) + pb) =
            f32_to_bf16_bits(p);
      }
      sm[n] += __shfl_xor(sm[n], 16, WAVE_SIZE);
      sm[n] += __shfl_xor(sm[n], 32, WAVE_SIZE);
    }
    if (lane < 32) wred[wid][lane] = (lane < 16) ? sm[0] : sm[1];
    __syncthreads();
    if (tid < QTILE)
      l_s[tid] = l_s[tid] * resc[tid] + wred[0][tid] + wred[1][tid] +
                 wred[2][tid] + wred[3][tid];

    // ---- O^T += V^T . P --------------------------------------------------------
#pragma unroll
    for (int mt = 0; mt < MT; ++mt)
#pragma unroll
      for (int nt = 0; nt < 2; ++nt) {
        const float r = resc[nt * 16 + l15];
#pragma unroll
        for (int rr = 0; rr < 4; ++rr) acc_o[mt][nt][rr] *= r;
      }
#pragma unroll
    for (int s = 0; s < KTILE / 32; ++s) {
#pragma unroll
      for (int mt = 0; mt < MT; ++mt) {
        const int drow = (wid * MT + mt) * 16 + l15;
        const int va = swz(drow * KTILE * 2 + s * 64 + l4 * 16, drow);
        const bf16x8v afrag = *reinterpret_cast<const bf16x8v*>(
            reinterpret_cast<const char*>(VTl) + va);
#pragma unroll
        for (int nt = 0; nt < 2; ++nt) {
          const int qrow_i = nt * 16 + l15;
          const int pb = swz(qrow_i * KTILE * 2 + s * 64 + l4 * 16, qrow_i);
          const bf16x8v bfrag = *reinterpret_cast<const bf16x8v*>(
              reinterpret_cast<const char*>(Pl) + pb);
          acc_o[mt][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag, bfrag, acc_o[mt][nt], 0, 0, 0);
        }
      }
    }
  }

  // ---- normalize + store O (O^T acc: col=q, row=d) ----------------------------
  __syncthreads();
#pragma unroll
  for (int nt = 0; nt < 2; ++nt) {
    const int qcol = nt * 16 + l15;
    if (r0 + qcol >= QL) continue;
    float l = l_s[qcol];
    if (sinks != nullptr) l += __expf(sinks[h] - m_s[qcol]);
    const float inv = 1.f / l;
#pragma unroll
    for (int mt = 0; mt < MT; ++mt) {
      const int d0 = (wid * MT + mt) * 16 + l4 * 4;
      uint16_t vals[4];
#pragma unroll
      for (int r = 0; r < 4; ++r)
        vals[r] = f32_to_bf16_bits(acc_o[mt][nt][r] * inv);
      *reinterpret_cast<uint2*>(
          out + ((size_t)(q0 + r0 + qcol) * Hq + h) * HEAD_DIM + d0) =
          *reinterpret_cast<const uint2*>(vals);
    }
  }
}

extern "C" void launch_prefill_attention(
    void* out, const void* q, const void* k_cache, const void* v_cache,
    const int* block_tables, const int* seq_lens, const int* cu_q,
    const int* tile_req, const int* tile_row0, int n_tiles, int Hq, int Hk,
    int D, int BS, int max_blocks, int64_t q_stride, float scale,
    int sliding_window, float softcap, const float* sinks, bool kv_fp8,
    float k_scale, float v_scale, hipStream_t stream, bool* launched) {
  *launched = false;
  dim3 grid(Hq, n_tiles, 1);
#define PF_LAUNCH(HD, FP8)                                                   \
  prefill_attention_kernel<HD, FP8><<<grid, PF_THREADS, 0, stream>>>(        \
      (uint16_t*)out, (const uint16_t*)q, k_cache, v_cache, block_tables,    \
      seq_lens, cu_q, tile_req, tile_row0, max_blocks, Hq, Hk, BS, q_stride, \
      scale, sliding_window, softcap, sinks, k_scale, v_scale);              \
  *launched = true;
  if (D == 128 && kv_fp8) { PF_LAUNCH(128, true) }
  else if (D == 128) { PF_LAUNCH(128, false) }
  else if (D == 64 && kv_fp8) { PF_LAUNCH(64, true) }
  else if (D == 64) { PF_LAUNCH(64, false) }
#undef PF_LAUNCH
}
