// Grouped MoE GEMMs for gfx950 — device-side routing, graph-capture safe.
//
// Pipeline (host side in ops/__init__.py, all capturable torch ops):
//   topk -> sort assignments by expert -> searchsorted segment offsets
// then three kernels here:
//   build_moe_tiles:  segment offsets -> (expert, row0) per 16-row tile
//   moe_gate_up:      h = act(x[gather] @ W_gu[e]^T)   (gate/up column pairs
//                     land in the same lane -> activation fused in-register)
//   moe_down:         out[t] += route_w * (h @ W_down[e]^T)  (fp32 atomics)
//
// Shapes: W_gu [E, 2I, H], W_down [E, H, I] (row-major, out-major) — B operand
// fragments read rows k-consecutive, so no transposition anywhere. Fixed
// MAX_TILES grid (tiles past the live count exit) keeps kernel launches
// hipGraph-stable. Reference analogue: the reference delegates MoE to
// SGLang/vLLM fused kernels; this is the MI355X-native equivalent.

#include "common.h"

#define MOE_THREADS 256
#define MOE_TM 64  // assignment rows per tile (4 MFMA M-tiles; big tiles
                   // amortize the expert-weight stream across more tokens)

__global__ void build_moe_tiles_kernel(
    int* __restrict__ tile_expert,   // [max_tiles]
    int* __restrict__ tile_row0,     // [max_tiles]
    const int* __restrict__ seg_offsets,  // [E+1]
    const int E, const int max_tiles) {
  if (threadIdx.x != 0 || blockIdx.x != 0) return;
  int t = 0;
  for (int e = 0; e < E && t < max_tiles; ++e) {
    const int begin = seg_offsets[e], end = seg_offsets[e + 1];
    for (int r = begin; r < end && t < max_tiles; r += MOE_TM) {
      tile_expert[t] = e;
      tile_row0[t] = r;
      ++t;
    }
  }
  for (; t < max_tiles; ++t) tile_expert[t] = -1;
}

// ---- grouped gate_up + activation --------------------------------------------
// grid: (max_tiles, I/64). Wave w handles N-tile w of a 64-col slab; each lane
// accumulates the gate tile and the matching up tile (same columns of I).

template <bool GELU>
__global__ __launch_bounds__(MOE_THREADS) void moe_gate_up_kernel(
    uint16_t* __restrict__ h_buf,          // [rows_padded, I] bf16
    const uint16_t* __restrict__ x,        // [T, H]
    const uint16_t* __restrict__ w_gu,     // [E, 2I, H]
    const int* __restrict__ tile_expert,
    const int* __restrict__ tile_row0,
    const int64_t* __restrict__ perm,      // [T*k] sorted-assignment -> flat idx
    const int* __restrict__ seg_offsets,   // [E+1]
    const int E, const int topk, const int H, const int I,
    const float limit,                     // >0: gpt-oss clamped act
    const uint16_t* __restrict__ bias_gu = nullptr) {  // [E, 2I] or null
  const int tile = blockIdx.x;
  const int e = tile_expert[tile];
  if (e < 0) return;
  const int row0 = tile_row0[tile];
  const int rows_end = seg_offsets[e + 1];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int l15 = lane & 15;
  const int l4 = lane >> 4;

  const int n0 = blockIdx.y * 64 + wid * 16;  // column tile within I
  // gate row n0+l15 of W_gu; up row I+n0+l15
  const uint16_t* wg_row = w_gu + ((size_t)e * 2 * I + (n0 + l15)) * H;
  const uint16_t* wu_row = w_gu + ((size_t)e * 2 * I + (I + n0 + l15)) * H;

  // 4 token M-tiles share each streamed W fragment
  const uint16_t* xrow[4];
  bool arow_ok[4];
#pragma unroll
  for (int mt = 0; mt < 4; ++mt) {
    const int arow = row0 + mt * 16 + l15;
    arow_ok[mt] = arow < rows_end;
    xrow[mt] = arow_ok[mt] ? x + (size_t)(perm[arow] / topk) * H : x;
  }

  f32x4v acc_g[4] = {}, acc_u[4] = {};
  for (int k = 0; k < H; k += 32) {
    const bf16x8v bg = *reinterpret_cast<const bf16x8v*>(wg_row + k + l4 * 8);
    const bf16x8v bu = *reinterpret_cast<const bf16x8v*>(wu_row + k + l4 * 8);
#pragma unroll
    for (int mt = 0; mt < 4; ++mt) {
      if (row0 + mt * 16 >= rows_end) break;
      bf16x8v afrag = {};
      if (arow_ok[mt])
        afrag = *reinterpret_cast<const bf16x8v*>(xrow[mt] + k + l4 * 8);
      acc_g[mt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bg, acc_g[mt], 0, 0, 0);
      acc_u[mt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bu, acc_u[mt], 0, 0, 0);
    }
  }

  // D layout (verified by the numerics tests): element (lane, reg) is
  // row m = (lane>>4)*4 + reg (token), col n = lane&15 (output feature).
#pragma unroll
  for (int mt = 0; mt < 4; ++mt) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int m = mt * 16 + l4 * 4 + r;  // token row within tile
      const int grow = row0 + m;
      if (grow >= rows_end) continue;
      const int n = n0 + l15;              // column within I
      float g = acc_g[mt][r], u = acc_u[mt][r];
      if (bias_gu != nullptr) {
        g += bf16_bits_to_f32(bias_gu[(size_t)e * 2 * I + n]);
        u += bf16_bits_to_f32(bias_gu[(size_t)e * 2 * I + I + n]);
      }
      float a;
      if (limit > 0.f) {                   // gpt-oss clamped act
        g = fminf(g, limit);
        u = fminf(fmaxf(u, -limit), limit);
        a = (u + 1.f) * (g / (1.f + __expf(-g * 1.702f)));
      } else if (GELU) {
        const float c = 0.7978845608028654f;
        a = 0.5f * g * (1.f + tanhf(c * (g + 0.044715f * g * g * g))) * u;
      } else {
        a = g / (1.f + __expf(-g)) * u;
      }
      h_buf[(size_t)grow * I + n] = f32_to_bf16_bits(a);
    }
  }
}

// ---- grouped down projection + weighted scatter -------------------------------
// grid: (max_tiles, H/64). out is fp32 [T, H]; routing weight applied here.

__global__ __launch_bounds__(MOE_THREADS) void moe_down_kernel(
    float* __restrict__ out,               // [T, H] fp32 (pre-zeroed)
    const uint16_t* __restrict__ h_buf,    // [rows_padded, I]
    const uint16_t* __restrict__ w_down,   // [E, H, I]
    const float* __restrict__ route_w,     // [T*k] flat routing weights
    const int* __restrict__ tile_expert,
    const int* __restrict__ tile_row0,
    const int64_t* __restrict__ perm,
    const int* __restrict__ seg_offsets,
    const int E, const int topk, const int H, const int I) {
  const int tile = blockIdx.x;
  const int e = tile_expert[tile];
  if (e < 0) return;
  const int row0 = tile_row0[tile];
  const int rows_end = seg_offsets[e + 1];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int l15 = lane & 15;
  const int l4 = lane >> 4;

  const int n0 = blockIdx.y * 256 + wid * 64;  // 4 waves x 4 n-tiles = 256 cols
  const uint16_t* hrow[4];
  bool arow_ok[4];
#pragma unroll
  for (int mt = 0; mt < 4; ++mt) {
    const int arow = row0 + mt * 16 + l15;
    arow_ok[mt] = arow < rows_end;
    hrow[mt] = arow_ok[mt] ? h_buf + (size_t)arow * I : h_buf;
  }

  f32x4v acc[4][4] = {};
  for (int k = 0; k < I; k += 32) {
    bf16x8v bfr[4];
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      const int n = n0 + nt * 16 + l15;
      bfr[nt] = {};
      if (n < H)
        bfr[nt] = *reinterpret_cast<const bf16x8v*>(
            w_down + ((size_t)e * H + n) * I + k + l4 * 8);
    }
#pragma unroll
    for (int mt = 0; mt < 4; ++mt) {
      if (row0 + mt * 16 >= rows_end) break;
      bf16x8v afrag = {};
      if (arow_ok[mt])
        afrag = *reinterpret_cast<const bf16x8v*>(hrow[mt] + k + l4 * 8);
#pragma unroll
      for (int nt = 0; nt < 4; ++nt)
        acc[mt][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afrag, bfr[nt], acc[mt][nt], 0, 0, 0);
    }
  }
#pragma unroll
  for (int mt = 0; mt < 4; ++mt) {
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      const int n = n0 + nt * 16 + l15;
      if (n >= H) break;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int grow = row0 + mt * 16 + l4 * 4 + r;
        if (grow >= rows_end) continue;
        const float w = route_w[grow];
        const int64_t token = perm[grow] / topk;
        atomicAdd(out + (size_t)token * H + n, acc[mt][nt][r] * w);
      }
    }
  }
}

// ---- fp8 (OCP e4m3) grouped path ---------------------------------------------
// W8A8: weights pre-quantized per output channel (w_scale[n] = amax_row/448,
// host side); activations quantized per token row on the fly. Non-scaled fp8
// MFMA runs at the bf16 rate on gfx950 (guide §3: only MX-block-scaled K=128
// reaches the 2x fp8 peak), so the win here is the HALVED weight stream —
// which is exactly what bounds grouped MoE: each expert's weights are read
// once per activated tile while tokens/expert stay small. Matches the
// BASELINE DeepSeek-V3 "fp8 MFMA" config.

#define FP8_MAX 448.f

__global__ __launch_bounds__(256) void quantize_fp8_rows_kernel(
    uint8_t* __restrict__ out,      // [R, C]
    float* __restrict__ scales,     // [R] dequant scale (amax/448)
    const uint16_t* __restrict__ x, // [R, C] bf16
    const int C) {
  const int row = blockIdx.x;
  const uint16_t* xr = x + (size_t)row * C;
  uint8_t* orow = out + (size_t)row * C;
  float amax = 1e-8f;
  for (int c = threadIdx.x * 8; c < C; c += blockDim.x * 8) {
    bf16x8 v = load_bf16x8(xr + c);
#pragma unroll
    for (int i = 0; i < 8; ++i) amax = fmaxf(amax, fabsf(bf16x8_get(v, i)));
  }
  __shared__ float lds[8];
  const int lane = threadIdx.x & 63, wid = threadIdx.x >> 6;
  amax = wave_reduce_max(amax);
  if (lane == 0) lds[wid] = amax;
  __syncthreads();
  amax = fmaxf(fmaxf(lds[0], lds[1]), fmaxf(lds[2], lds[3]));
  if (threadIdx.x == 0) scales[row] = amax / FP8_MAX;
  const float inv = FP8_MAX / amax;
  for (int c = threadIdx.x * 8; c < C; c += blockDim.x * 8) {
    bf16x8 v = load_bf16x8(xr + c);
    uint64_t qv = 0;
#pragma unroll
    for (int i = 0; i < 8; ++i)
      qv |= (uint64_t)f32_to_fp8_e4m3(bf16x8_get(v, i) * inv) << (8 * i);
    *reinterpret_cast<uint64_t*>(orow + c) = qv;
  }
}

// grid: (max_tiles, I/64); acc dequant = x_scale[token] * w_scale[n].
template <bool GELU>
__global__ __launch_bounds__(MOE_THREADS) void moe_gate_up_fp8_kernel(
    uint16_t* __restrict__ h_buf,          // [rows_padded, I] bf16
    const uint8_t* __restrict__ x,         // [T, H] fp8
    const float* __restrict__ x_scale,     // [T]
    const uint8_t* __restrict__ w_gu,      // [E, 2I, H] fp8
    const float* __restrict__ w_scale,     // [E, 2I]
    const int* __restrict__ tile_expert,
    const int* __restrict__ tile_row0,
    const int64_t* __restrict__ perm,
    const int* __restrict__ seg_offsets,
    const int E, const int topk, const int H, const int I,
    const float limit) {
  const int tile = blockIdx.x;
  const int e = tile_expert[tile];
  if (e < 0) return;
  const int row0 = tile_row0[tile];
  const int rows_end = seg_offsets[e + 1];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int l15 = lane & 15;
  const int l4 = lane >> 4;

  const int n0 = blockIdx.y * 64 + wid * 16;
  const uint8_t* wg_row = w_gu + ((size_t)e * 2 * I + (n0 + l15)) * H;
  const uint8_t* wu_row = w_gu + ((size_t)e * 2 * I + (I + n0 + l15)) * H;
  const float sg = w_scale[(size_t)e * 2 * I + n0 + l15];
  const float su = w_scale[(size_t)e * 2 * I + I + n0 + l15];

  const uint8_t* xrow[4];
  bool arow_ok[4];
#pragma unroll
  for (int mt = 0; mt < 4; ++mt) {
    const int arow = row0 + mt * 16 + l15;
    arow_ok[mt] = arow < rows_end;
    xrow[mt] = arow_ok[mt] ? x + (size_t)(perm[arow] / topk) * H : x;
  }

  f32x4v acc_g[4] = {}, acc_u[4] = {};
  for (int k = 0; k < H; k += 32) {
    const long bg = *reinterpret_cast<const long*>(wg_row + k + l4 * 8);
    const long bu = *reinterpret_cast<const long*>(wu_row + k + l4 * 8);
#pragma unroll
    for (int mt = 0; mt < 4; ++mt) {
      if (row0 + mt * 16 >= rows_end) break;
      long afrag = 0;
      if (arow_ok[mt])
        afrag = *reinterpret_cast<const long*>(xrow[mt] + k + l4 * 8);
      acc_g[mt] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(afrag, bg, acc_g[mt], 0, 0, 0);
      acc_u[mt] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(afrag, bu, acc_u[mt], 0, 0, 0);
    }
  }

#pragma unroll
  for (int mt = 0; mt < 4; ++mt) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int m = mt * 16 + l4 * 4 + r;
      const int grow = row0 + m;
      if (grow >= rows_end) continue;
      const int n = n0 + l15;
      const float sx = x_scale[perm[grow] / topk];
      float g = acc_g[mt][r] * sx * sg, u = acc_u[mt][r] * sx * su;
      float a;
      if (limit > 0.f) {
        g = fminf(g, limit);
        u = fminf(fmaxf(u, -limit), limit);
        a = (u + 1.f) * (g / (1.f + __expf(-g * 1.702f)));
      } else if (GELU) {
        const float c = 0.7978845608028654f;
        a = 0.5f * g * (1.f + tanhf(c * (g + 0.044715f * g * g * g))) * u;
      } else {
        a = g / (1.f + __expf(-g)) * u;
      }
      h_buf[(size_t)grow * I + n] = f32_to_bf16_bits(a);
    }
  }
}

// grid: (max_tiles, H/256); h rows quantized by a quantize_fp8_rows pass.
__global__ __launch_bounds__(MOE_THREADS) void moe_down_fp8_kernel(
    float* __restrict__ out,               // [T, H] fp32 (pre-zeroed)
    const uint8_t* __restrict__ h_fp8,     // [rows_padded, I]
    const float* __restrict__ h_scale,     // [rows_padded]
    const uint8_t* __restrict__ w_down,    // [E, H, I] fp8
    const float* __restrict__ w_scale,     // [E, H]
    const float* __restrict__ route_w,
    const int* __restrict__ tile_expert,
    const int* __restrict__ tile_row0,
    const int64_t* __restrict__ perm,
    const int* __restrict__ seg_offsets,
    const int E, const int topk, const int H, const int I) {
  const int tile = blockIdx.x;
  const int e = tile_expert[tile];
  if (e < 0) return;
  const int row0 = tile_row0[tile];
  const int rows_end = seg_offsets[e + 1];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int l15 = lane & 15;
  const int l4 = lane >> 4;

  const int n0 = blockIdx.y * 256 + wid * 64;
  const uint8_t* hrow[4];
  bool arow_ok[4];
#pragma unroll
  for (int mt = 0; mt < 4; ++mt) {
    const int arow = row0 + mt * 16 + l15;
    arow_ok[mt] = arow < rows_end;
    hrow[mt] = arow_ok[mt] ? h_fp8 + (size_t)arow * I : h_fp8;
  }

  f32x4v acc[4][4] = {};
  for (int k = 0; k < I; k += 32) {
    long bfr[4];
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      const int n = n0 + nt * 16 + l15;
      bfr[nt] = 0;
      if (n < H)
        bfr[nt] = *reinterpret_cast<const long*>(
            w_down + ((size_t)e * H + n) * I + k + l4 * 8);
    }
#pragma unroll
    for (int mt = 0; mt < 4; ++mt) {
      if (row0 + mt * 16 >= rows_end) break;
      long afrag = 0;
      if (arow_ok[mt])
        afrag = *reinterpret_cast<const long*>(hrow[mt] + k + l4 * 8);
#pragma unroll
      for (int nt = 0; nt < 4; ++nt)
        acc[mt][nt] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(
            afrag, bfr[nt], acc[mt][nt], 0, 0, 0);
    }
  }
#pragma unroll
  for (int mt = 0; mt < 4; ++mt) {
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      const int n = n0 + nt * 16 + l15;
      if (n >= H) break;
      const float sw = w_scale[(size_t)e * H + n];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int grow = row0 + mt * 16 + l4 * 4 + r;
        if (grow >= rows_end) continue;
        const float w = route_w[grow] * h_scale[grow] * sw;
        const int64_t token = perm[grow] / topk;
        atomicAdd(out + (size_t)token * H + n, acc[mt][nt][r] * w);
      }
    }
  }
}

extern "C" void launch_quantize_fp8_rows(
    void* out, float* scales, const void* x, int R, int C,
    hipStream_t stream) {
  quantize_fp8_rows_kernel<<<R, 256, 0, stream>>>(
      (uint8_t*)out, scales, (const uint16_t*)x, C);
}

extern "C" void launch_moe_gate_up_fp8(
    void* h_buf, const void* x, const float* x_scale, const void* w_gu,
    const float* w_scale, const int* tile_expert, const int* tile_row0,
    const int64_t* perm, const int* seg_offsets, int E, int topk, int H, int I,
    int max_tiles, bool gelu, float limit, hipStream_t stream) {
  dim3 grid(max_tiles, ceil_div(I, 64), 1);
  if (gelu)
    moe_gate_up_fp8_kernel<true><<<grid, MOE_THREADS, 0, stream>>>(
        (uint16_t*)h_buf, (const uint8_t*)x, x_scale, (const uint8_t*)w_gu,
        w_scale, tile_expert, tile_row0, perm, seg_offsets, E, topk, H, I,
        limit);
  else
    moe_gate_up_fp8_kernel<false><<<grid, MOE_THREADS, 0, stream>>>(
        (uint16_t*)h_buf, (const uint8_t*)x, x_scale, (const uint8_t*)w_gu,
        w_scale, tile_expert, tile_row0, perm, seg_offsets, E, topk, H, I,
        limit);
}

extern "C" void launch_moe_down_fp8(
    void* out, const void* h_fp8, const float* h_scale, const void* w_down,
    const float* w_scale, const float* route_w, const int* tile_expert,
    const int* tile_row0, const int64_t* perm, const int* seg_offsets, int E,
    int topk, int H, int I, int max_tiles, hipStream_t stream) {
  dim3 grid(max_tiles, ceil_div(H, 256), 1);
  moe_down_fp8_kernel<<<grid, MOE_THREADS, 0, stream>>>(
      (float*)out, (const uint8_t*)h_fp8, h_scale, (const uint8_t*)w_down,
      w_scale, route_w, tile_expert, tile_row0, perm, seg_offsets, E, topk, H,
      I);
}

extern "C" void launch_build_moe_tiles(
    int* tile_expert, int* tile_row0, const int* seg_offsets, int E,
    int max_tiles, hipStream_t stream) {
  build_moe_tiles_kernel<<<1, 64, 0, stream>>>(tile_expert, tile_row0,
                                               seg_offsets, E, max_tiles);
}

extern "C" void launch_moe_gate_up(
    void* h_buf, const void* x, const void* w_gu, const int* tile_expert,
    const int* tile_row0, const int64_t* perm, const int* seg_offsets, int E,
    int topk, int H, int I, int max_tiles, bool gelu, float limit,
    const void* bias_gu, hipStream_t stream) {
  dim3 grid(max_tiles, ceil_div(I, 64), 1);
  if (gelu)
    moe_gate_up_kernel<true><<<grid, MOE_THREADS, 0, stream>>>(
        (uint16_t*)h_buf, (const uint16_t*)x, (const uint16_t*)w_gu,
        tile_expert, tile_row0, perm, seg_offsets, E, topk, H, I, limit,
        (const uint16_t*)bias_gu);
  else
    moe_gate_up_kernel<false><<<grid, MOE_THREADS, 0, stream>>>(
        (uint16_t*)h_buf, (const uint16_t*)x, (const uint16_t*)w_gu,
        tile_expert, tile_row0, perm, seg_offsets, E, topk, H, I, limit,
        (const uint16_t*)bias_gu);
}

extern "C" void launch_moe_down(
    void* out, const void* h_buf, const void* w_down, const float* route_w,
    const int* tile_expert, const int* tile_row0, const int64_t* perm,
    const int* seg_offsets, int E, int topk, int H, int I, int max_tiles,
    hipStream_t stream) {
  dim3 grid(max_tiles, ceil_div(H, 256), 1);
  moe_down_kernel<<<grid, MOE_THREADS, 0, stream>>>(
      (float*)out, (const uint16_t*)h_buf, (const uint16_t*)w_down, route_w,
      tile_expert, tile_row0, perm, seg_offsets, E, topk, H, I);
}
