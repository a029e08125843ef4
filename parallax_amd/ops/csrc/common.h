// Common device helpers for the parallax_amd CDNA4 (gfx950) kernels.
// Wave size is 64 on CDNA; all reductions and masks below assume that.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <hip/hip_fp8.h>
#include <stdint.h>

#define WAVE_SIZE 64

#define DEVINL __device__ __forceinline__

// ---- packed bf16 helpers -----------------------------------------------------
// hipcc does not auto-vectorize scalar bf16 loads (guide G13); we move 16 B/lane
// as int4 and unpack.

struct bf16x8 {
  int4 raw;  // 8 bf16 values
};

DEVINL float bf16_bits_to_f32(uint16_t h) {
  union {
    float f;
    uint32_t u;
  } v;
  v.u = static_cast<uint32_t>(h) << 16;
  return v.f;
}

DEVINL uint16_t f32_to_bf16_bits(float f) {
  union {
    float f;
    uint32_t u;
  } v;
  v.f = f;
  // round-to-nearest-even
  uint32_t rounding = 0x7fff + ((v.u >> 16) & 1);
  return static_cast<uint16_t>((v.u + rounding) >> 16);
}

// unpack the i-th bf16 (i in [0,8)) of a 16-byte chunk
DEVINL float bf16x8_get(const bf16x8& p, int i) {
  const uint32_t* u = reinterpret_cast<const uint32_t*>(&p.raw);
  uint32_t w = u[i >> 1];
  uint16_t h = (i & 1) ? static_cast<uint16_t>(w >> 16) : static_cast<uint16_t>(w & 0xffff);
  return bf16_bits_to_f32(h);
}

DEVINL bf16x8 load_bf16x8(const void* ptr) {
  bf16x8 out;
  out.raw = *reinterpret_cast<const int4*>(ptr);
  return out;
}

DEVINL void store_bf16x8(void* ptr, const float* vals) {
  uint32_t u[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    uint32_t lo = f32_to_bf16_bits(vals[2 * i]);
    uint32_t hi = f32_to_bf16_bits(vals[2 * i + 1]);
    u[i] = lo | (hi << 16);
  }
  *reinterpret_cast<int4*>(ptr) = *reinterpret_cast<const int4*>(u);
}

// dot of 8 packed bf16 with 8 fp32 values
DEVINL float bf16x8_dot(const bf16x8& k, const float* q) {
  float acc = 0.f;
#pragma unroll
  for (int i = 0; i < 8; ++i) acc = fmaf(bf16x8_get(k, i), q[i], acc);
  return acc;
}

// ---- wave reductions ---------------------------------------------------------

DEVINL float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE_SIZE);
  return v;
}

DEVINL float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, WAVE_SIZE));
  return v;
}

// block reduction via LDS (blockDim.x threads, up to 1024)
template <int MAX_WAVES = 16>
DEVINL float block_reduce_sum(float v, float* lds_scratch) {
  int lane = threadIdx.x & (WAVE_SIZE - 1);
  int wid = threadIdx.x / WAVE_SIZE;
  v = wave_reduce_sum(v);
  if (lane == 0) lds_scratch[wid] = v;
  __syncthreads();
  int nw = (blockDim.x + WAVE_SIZE - 1) / WAVE_SIZE;
  v = (threadIdx.x < (unsigned)nw) ? lds_scratch[threadIdx.x] : 0.f;
  if (wid == 0) {
#pragma unroll
    for (int off = MAX_WAVES / 2; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE_SIZE);
    if (lane == 0) lds_scratch[0] = v;
  }
  __syncthreads();
  return lds_scratch[0];
}

// ---- FP8 (OCP e4m3fn — gfx950 native; NOT the MI300X fnuz variant) -----------

DEVINL float fp8_e4m3_to_f32(uint8_t b) {
  __hip_fp8_e4m3 v;
  v.__x = b;
  return static_cast<float>(v);
}

typedef float f32x2v __attribute__((ext_vector_type(2)));

// 8 fp8-E4M3 bytes -> 8 bf16 (scaled): packed HW converts
// (v_cvt_pk_f32_fp8 on dword halves) instead of 8 byte-extract + scalar
// convert chains — the fp8-KV fragment-load path was VALU-bound on the
// scalar version (profiles/README.md round 2 fp8-KV note).
DEVINL void fp8x8_to_bf16x8(uint64_t raw, float scale, uint16_t* out8) {
  const uint32_t lo = (uint32_t)raw;
  const uint32_t hi = (uint32_t)(raw >> 32);
  f32x2v f[4];
  f[0] = __builtin_amdgcn_cvt_pk_f32_fp8(lo, false);
  f[1] = __builtin_amdgcn_cvt_pk_f32_fp8(lo, true);
  f[2] = __builtin_amdgcn_cvt_pk_f32_fp8(hi, false);
  f[3] = __builtin_amdgcn_cvt_pk_f32_fp8(hi, true);
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    out8[2 * i] = f32_to_bf16_bits(f[i][0] * scale);
    out8[2 * i + 1] = f32_to_bf16_bits(f[i][1] * scale);
  }
}

DEVINL uint8_t f32_to_fp8_e4m3(float f) {
  __hip_fp8_e4m3 v(f);
  return v.__x;
}

typedef short bf16x8v __attribute__((ext_vector_type(8)));
typedef float f32x4v __attribute__((ext_vector_type(4)));

// XOR-swizzle a byte offset within a row-major LDS tile: spreads the 16-lane
// row-stride access pattern across banks (guide §6 G4)
DEVINL int swz(int byte_in, int row) { return byte_in ^ ((row & 7) << 4); }

__host__ __device__ __forceinline__ int ceil_div(int a, int b) {
  return (a + b - 1) / b;
}

__host__ __forceinline__ int clamp_int(long long v, int lo, int hi) {
  if (v < lo) return lo;
  if (v > hi) return hi;
  return (int)v;
}
