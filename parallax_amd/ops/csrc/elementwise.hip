// Memory-bound elementwise / normalization kernels for gfx950.
// All bf16 traffic is vectorized 16 B/lane (bf16x8) per the CDNA guide (G13).
//
// Reference analogues: RMSNorm/RoPE/activation live inside MLX / SGLang in the
// reference; here they are first-class HIP kernels fused where it pays
// (fused_add_rmsnorm saves two HBM round-trips per layer).

#include "common.h"

// ---- rmsnorm -------------------------------------------------------------------
// x: [rows, H] bf16, w: [H] bf16, out: [rows, H]. One block per row.
// fp32 accumulation; H must be a multiple of 8.

template <bool FUSED_ADD>
__global__ void rmsnorm_kernel(
    uint16_t* __restrict__ out,          // [rows, H] (bf16 bits)
    uint16_t* __restrict__ x,            // [rows, H] — input; mutated if FUSED_ADD
    uint16_t* __restrict__ residual,     // [rows, H] or nullptr
    const uint16_t* __restrict__ w,      // [H]
    const float eps,
    const int H) {
  __shared__ float red[16];
  const int row = blockIdx.x;
  uint16_t* xr = x + (size_t)row * H;
  uint16_t* rr = FUSED_ADD ? residual + (size_t)row * H : nullptr;
  uint16_t* orow = out + (size_t)row * H;

  const int vecs = H / 8;
  float ss = 0.f;
  // pass 1: (optional add) + sum of squares; FUSED_ADD writes the new residual
  for (int i = threadIdx.x; i < vecs; i += blockDim.x) {
    bf16x8 xv = load_bf16x8(xr + i * 8);
    float vals[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) vals[j] = bf16x8_get(xv, j);
    if (FUSED_ADD) {
      bf16x8 rv = load_bf16x8(rr + i * 8);
#pragma unroll
      for (int j = 0; j < 8; ++j) vals[j] += bf16x8_get(rv, j);
      store_bf16x8(rr + i * 8, vals);  // residual' = x + residual
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) ss = fmaf(vals[j], vals[j], ss);
  }
  ss = block_reduce_sum(ss, red);
  const float rstd = rsqrtf(ss / H + eps);
  __syncthreads();

  // pass 2: normalize * weight  (rows are L2-resident between passes)
  const uint16_t* src = FUSED_ADD ? rr : xr;
  for (int i = threadIdx.x; i < vecs; i += blockDim.x) {
    bf16x8 xv = load_bf16x8(src + i * 8);
    bf16x8 wv = load_bf16x8(w + i * 8);
    float vals[8];
#pragma unroll
    for (int j = 0; j < 8; ++j)
      vals[j] = bf16x8_get(xv, j) * rstd * bf16x8_get(wv, j);
    store_bf16x8(orow + i * 8, vals);
  }
}

extern "C" void launch_rmsnorm(
    void* out, void* x, const void* w, float eps, int rows, int H,
    hipStream_t stream) {
  int threads = min(1024, max(64, ((H / 8) + 63) / 64 * 64));
  rmsnorm_kernel<false><<<rows, threads, 0, stream>>>(
      (uint16_t*)out, (uint16_t*)x, nullptr, (const uint16_t*)w, eps, H);
}

extern "C" void launch_fused_add_rmsnorm(
    void* x, void* residual, const void* w, float eps, int rows, int H,
    hipStream_t stream) {
  int threads = min(1024, max(64, ((H / 8) + 63) / 64 * 64));
  // out == x (in-place): x <- rmsnorm(x + residual), residual <- x + residual
  rmsnorm_kernel<true><<<rows, threads, 0, stream>>>(
      (uint16_t*)x, (uint16_t*)x, (uint16_t*)residual, (const uint16_t*)w, eps, H);
}

// ---- RoPE (neox + interleaved) ---------------------------------------------------
// q: [T, Hq, D], k: [T, Hk, D] (bf16, mutated in place), positions: [T] i32,
// cos_sin: [P, rot] fp32 (cos | sin halves). Host-precomputed table (guide §B).
// One block per token; threads sweep (head, pair).

template <bool NEOX>
__global__ void rope_kernel(
    uint16_t* __restrict__ q, uint16_t* __restrict__ k,
    const int* __restrict__ positions, const float* __restrict__ cos_sin,
    const int Hq, const int Hk, const int D, const int rot,
    const int64_t q_stride, const int64_t k_stride) {
  const int t = blockIdx.x;
  const int half = rot / 2;
  const float* cs = cos_sin + (size_t)positions[t] * rot;
  const int total = (Hq + Hk) * half;
  for (int i = threadIdx.x; i < total; i += blockDim.x) {
    const int h = i / half;
    const int j = i % half;
    uint16_t* base = (h < Hq) ? q + (size_t)t * q_stride + h * D
                              : k + (size_t)t * k_stride + (h - Hq) * D;
    const float c = cs[j], s = cs[half + j];
    int i1, i2;
    if (NEOX) {
      i1 = j; i2 = j + half;
    } else {
      i1 = 2 * j; i2 = 2 * j + 1;
    }
    const float x1 = bf16_bits_to_f32(base[i1]);
    const float x2 = bf16_bits_to_f32(base[i2]);
    base[i1] = f32_to_bf16_bits(fmaf(x1, c, -x2 * s));
    base[i2] = f32_to_bf16_bits(fmaf(x2, c, x1 * s));
  }
}

extern "C" void launch_rope(
    void* q, void* k, const int* positions, const float* cos_sin,
    int T, int Hq, int Hk, int D, int rot, bool neox, int64_t q_stride,
    int64_t k_stride, hipStream_t stream) {
  int threads = min(512, max(64, ceil_div((Hq + Hk) * rot / 2, 64) * 64));
  if (neox)
    rope_kernel<true><<<T, threads, 0, stream>>>(
        (uint16_t*)q, (uint16_t*)k, positions, cos_sin, Hq, Hk, D, rot,
        q_stride, k_stride);
  else
    rope_kernel<false><<<T, threads, 0, stream>>>(
        (uint16_t*)q, (uint16_t*)k, positions, cos_sin, Hq, Hk, D, rot,
        q_stride, k_stride);
}

// ---- fused rope + cache scatter ----------------------------------------------------
// One launch per decode/prefill step replaces {rope(q,k) ; reshape_and_cache}:
// q roped in place (strided rows OK), k roped straight into the paged cache,
// v copied vectorized. Saves one k round-trip and two kernel launches per layer.
template <bool NEOX, bool FP8>
__global__ void rope_and_cache_kernel(
    uint16_t* __restrict__ q,        // [T, Hq, D] rows at q_stride
    const uint16_t* __restrict__ k,  // [T, Hk, D] rows at k_stride
    const uint16_t* __restrict__ v,
    void* __restrict__ k_cache_v,    // [NB, Hk, BS, D] bf16 | fp8
    void* __restrict__ v_cache_v,
    const int* __restrict__ positions, const float* __restrict__ cos_sin,
    const int64_t* __restrict__ slot_mapping,
    const int Hq, const int Hk, const int D, const int rot, const int BS,
    const int64_t q_stride, const int64_t k_stride, const int64_t v_stride,
    const float inv_k_scale, const float inv_v_scale) {
  const int t = blockIdx.x;
  const int64_t slot = slot_mapping[t];
  const int half = rot / 2;
  const float* cs = cos_sin + (size_t)positions[t] * rot;
  const int64_t blk = slot >= 0 ? slot / BS : 0;
  const int64_t off = slot >= 0 ? slot % BS : 0;
  const size_t base_off = (((size_t)blk * Hk) * BS + off) * D;
  // v cache is transposed [NB, Hk, D, BS]: per-(h,d) element lands at column
  // `off` of the d row
  const size_t vbase_off = ((size_t)blk * Hk) * (size_t)D * BS + off;
  uint16_t* kdst16 = FP8 ? nullptr : (uint16_t*)k_cache_v + base_off;
  uint16_t* vdst16 = FP8 ? nullptr : (uint16_t*)v_cache_v + vbase_off;
  uint8_t* kdst8 = FP8 ? (uint8_t*)k_cache_v + base_off : nullptr;
  uint8_t* vdst8 = FP8 ? (uint8_t*)v_cache_v + vbase_off : nullptr;

  // q: rope in place
  for (int i = threadIdx.x; i < Hq * half; i += blockDim.x) {
    const int h = i / half, j = i % half;
    uint16_t* base = q + (size_t)t * q_stride + h * D;
    const float c = cs[j], ss = cs[half + j];
    const int i1 = NEOX ? j : 2 * j;
    const int i2 = NEOX ? j + half : 2 * j + 1;
    const float x1 = bf16_bits_to_f32(base[i1]);
    const float x2 = bf16_bits_to_f32(base[i2]);
    base[i1] = f32_to_bf16_bits(fmaf(x1, c, -x2 * ss));
    base[i2] = f32_to_bf16_bits(fmaf(x2, c, x1 * ss));
  }
  if (slot < 0) return;
  // k: rope -> cache (rot dims) + copy (pass-through dims if rot < D)
  for (int i = threadIdx.x; i < Hk * half; i += blockDim.x) {
    const int h = i / half, j = i % half;
    const uint16_t* src = k + (size_t)t * k_stride + h * D;
    const float c = cs[j], ss = cs[half + j];
    const int i1 = NEOX ? j : 2 * j;
    const int i2 = NEOX ? j + half : 2 * j + 1;
    const float x1 = bf16_bits_to_f32(src[i1]);
    const float x2 = bf16_bits_to_f32(src[i2]);
    const float o1 = fmaf(x1, c, -x2 * ss);
    const float o2 = fmaf(x2, c, x1 * ss);
    if (FP8) {
      kdst8[(size_t)h * BS * D + i1] = f32_to_fp8_e4m3(o1 * inv_k_scale);
      kdst8[(size_t)h * BS * D + i2] = f32_to_fp8_e4m3(o2 * inv_k_scale);
    } else {
      kdst16[(size_t)h * BS * D + i1] = f32_to_bf16_bits(o1);
      kdst16[(size_t)h * BS * D + i2] = f32_to_bf16_bits(o2);
    }
  }
  if (rot < D) {
    for (int i = threadIdx.x; i < Hk * (D - rot); i += blockDim.x) {
      const int h = i / (D - rot);
      const int d = rot + i % (D - rot);
      const float x = bf16_bits_to_f32(k[(size_t)t * k_stride + h * D + d]);
      if (FP8)
        kdst8[(size_t)h * BS * D + d] = f32_to_fp8_e4m3(x * inv_k_scale);
      else
        kdst16[(size_t)h * BS * D + d] = f32_to_bf16_bits(x);
    }
  }
  // v: transpose-scatter into the [.., D, BS] layout (quantized when FP8) —
  // one 2 B store per element at decode (1 token); the attention kernels'
  // fragment loads read V^T contiguously in exchange
  for (int i = threadIdx.x; i < Hk * D; i += blockDim.x) {
    const int h = i / D;
    const int d = i % D;
    const float x = bf16_bits_to_f32(v[(size_t)t * v_stride + h * D + d]);
    const size_t dst = ((size_t)h * D + d) * BS;
    if (FP8)
      vdst8[dst] = f32_to_fp8_e4m3(x * inv_v_scale);
    else
      vdst16[dst] = f32_to_bf16_bits(x);
  }
}

extern "C" void launch_rope_and_cache(
    void* q, const void* k, const void* v, void* k_cache, void* v_cache,
    const int* positions, const float* cos_sin, const int64_t* slot_mapping,
    int T, int Hq, int Hk, int D, int rot, int BS, bool neox, bool fp8,
    float k_scale, float v_scale, int64_t q_stride, int64_t k_stride,
    int64_t v_stride, hipStream_t stream) {
  int threads = min(512, max(128, ceil_div(Hk * D / 8, 64) * 64));
  const float iks = 1.f / k_scale, ivs = 1.f / v_scale;
#define RC_LAUNCH(NEOX, FP8)                                                 \
  rope_and_cache_kernel<NEOX, FP8><<<T, threads, 0, stream>>>(              \
      (uint16_t*)q, (const uint16_t*)k, (const uint16_t*)v, k_cache,        \
      v_cache, positions, cos_sin, slot_mapping, Hq, Hk, D, rot, BS,        \
      q_stride, k_stride, v_stride, iks, ivs)
  if (neox && fp8) RC_LAUNCH(true, true);
  else if (neox) RC_LAUNCH(true, false);
  else if (fp8) RC_LAUNCH(false, true);
  else RC_LAUNCH(false, false);
#undef RC_LAUNCH
}

// ---- reshape_and_cache ------------------------------------------------------------
// k/v: [T, Hk, D] bf16 -> k_cache/v_cache [NB, Hk, BS, D] via slot_mapping [T]
// (slot = block * BS + off; -1 skips). One block per token, vectorized rows.

__global__ void reshape_and_cache_kernel(
    const uint16_t* __restrict__ k, const uint16_t* __restrict__ v,
    uint16_t* __restrict__ k_cache, uint16_t* __restrict__ v_cache,
    const int64_t* __restrict__ slot_mapping,
    const int Hk, const int D, const int BS) {
  const int t = blockIdx.x;
  const int64_t slot = slot_mapping[t];
  if (slot < 0) return;
  const int64_t blk = slot / BS, off = slot % BS;
  const int vecs = Hk * D / 8;
  const uint16_t* ksrc = k + (size_t)t * Hk * D;
  const uint16_t* vsrc = v + (size_t)t * Hk * D;
  for (int i = threadIdx.x; i < vecs; i += blockDim.x) {
    const int h = (i * 8) / D;
    const int d = (i * 8) % D;
    // k row: cache[blk][h][off][d]
    const size_t dst = (((size_t)blk * Hk + h) * BS + off) * D + d;
    *reinterpret_cast<int4*>(k_cache + dst) =
        *reinterpret_cast<const int4*>(ksrc + i * 8);
    // v transposed: cache[blk][h][d + j][off]
    const int4 vv = *reinterpret_cast<const int4*>(vsrc + i * 8);
    const uint16_t* ve = reinterpret_cast<const uint16_t*>(&vv);
#pragma unroll
    for (int j = 0; j < 8; ++j)
      v_cache[(((size_t)blk * Hk + h) * D + d + j) * BS + off] = ve[j];
  }
}

extern "C" void launch_reshape_and_cache(
    const void* k, const void* v, void* k_cache, void* v_cache,
    const int64_t* slot_mapping, int T, int Hk, int D, int BS,
    hipStream_t stream) {
  int threads = min(512, max(64, ceil_div(Hk * D / 8, 64) * 64));
  reshape_and_cache_kernel<<<T, threads, 0, stream>>>(
      (const uint16_t*)k, (const uint16_t*)v, (uint16_t*)k_cache,
      (uint16_t*)v_cache, slot_mapping, Hk, D, BS);
}

// MLA variant: latent [T, R] + rope [T, dr] -> cache [NB, BS, R+dr]
template <bool FP8>
__global__ void mla_reshape_and_cache_kernel(
    const uint16_t* __restrict__ latent, const uint16_t* __restrict__ k_rope,
    void* __restrict__ cache, const int64_t* __restrict__ slot_mapping,
    const int R, const int DR, const int BS, const float inv_scale) {
  const int t = blockIdx.x;
  const int64_t slot = slot_mapping[t];
  if (slot < 0) return;
  const int64_t blk = slot / BS, off = slot % BS;
  const size_t row = ((size_t)blk * BS + off) * (R + DR);
  if (FP8) {
    uint8_t* dst = (uint8_t*)cache + row;
    for (int i = threadIdx.x * 8; i < R; i += blockDim.x * 8) {
      uint64_t packed = 0;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        packed |= (uint64_t)f32_to_fp8_e4m3(
                      bf16_bits_to_f32(latent[(size_t)t * R + i + j]) *
                      inv_scale)
                  << (8 * j);
      *reinterpret_cast<uint64_t*>(dst + i) = packed;
    }
    for (int i = threadIdx.x * 8; i < DR; i += blockDim.x * 8) {
      uint64_t packed = 0;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        packed |= (uint64_t)f32_to_fp8_e4m3(
                      bf16_bits_to_f32(k_rope[(size_t)t * DR + i + j]) *
                      inv_scale)
                  << (8 * j);
      *reinterpret_cast<uint64_t*>(dst + R + i) = packed;
    }
  } else {
    uint16_t* dst = (uint16_t*)cache + row;
    for (int i = threadIdx.x * 8; i < R; i += blockDim.x * 8)
      *reinterpret_cast<int4*>(dst + i) =
          *reinterpret_cast<const int4*>(latent + (size_t)t * R + i);
    for (int i = threadIdx.x * 8; i < DR; i += blockDim.x * 8)
      *reinterpret_cast<int4*>(dst + R + i) =
          *reinterpret_cast<const int4*>(k_rope + (size_t)t * DR + i);
  }
}

extern "C" void launch_mla_reshape_and_cache(
    const void* latent, const void* k_rope, void* cache,
    const int64_t* slot_mapping, int T, int R, int DR, int BS, bool fp8,
    float inv_scale, hipStream_t stream) {
  if (fp8)
    mla_reshape_and_cache_kernel<true><<<T, 128, 0, stream>>>(
        (const uint16_t*)latent, (const uint16_t*)k_rope, cache, slot_mapping,
        R, DR, BS, inv_scale);
  else
    mla_reshape_and_cache_kernel<false><<<T, 128, 0, stream>>>(
        (const uint16_t*)latent, (const uint16_t*)k_rope, cache, slot_mapping,
        R, DR, BS, inv_scale);
}

// ---- activations ----------------------------------------------------------------------
// x: [T, 2I] (gate | up) -> out: [T, I];  grid-stride, vectorized.

template <bool GELU>
__global__ void act_and_mul_kernel(
    uint16_t* __restrict__ out, const uint16_t* __restrict__ x,
    const int64_t T, const int I) {
  const int64_t total = T * (I / 8);
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (int64_t)gridDim.x * blockDim.x) {
    const int64_t t = idx / (I / 8);
    const int i = (idx % (I / 8)) * 8;
    bf16x8 g = load_bf16x8(x + t * 2 * I + i);
    bf16x8 u = load_bf16x8(x + t * 2 * I + I + i);
    float vals[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float gv = bf16x8_get(g, j);
      const float uv = bf16x8_get(u, j);
      float a;
      if (GELU) {  // tanh approximation
        const float c = 0.7978845608028654f;
        a = 0.5f * gv * (1.f + tanhf(c * (gv + 0.044715f * gv * gv * gv)));
      } else {  // silu
        a = gv / (1.f + __expf(-gv));
      }
      vals[j] = a * uv;
    }
    store_bf16x8(out + t * I + i, vals);
  }
}

extern "C" void launch_act_and_mul(
    void* out, const void* x, int64_t T, int I, bool gelu, hipStream_t stream) {
  long long total = (long long)T * (I / 8);
  int blocks = clamp_int((total + 255) / 256, 1, 2048);
  if (gelu)
    act_and_mul_kernel<true><<<blocks, 256, 0, stream>>>(
        (uint16_t*)out, (const uint16_t*)x, T, I);
  else
    act_and_mul_kernel<false><<<blocks, 256, 0, stream>>>(
        (uint16_t*)out, (const uint16_t*)x, T, I);
}
