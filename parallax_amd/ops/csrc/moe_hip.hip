#include "hip/hip_runtime.h"
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
    const float limit) {                   // >0: gpt-oss clamped act
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

extern "C" void launch_build_moe_tiles(
    int* tile_expert, int* tile_row0, const int* seg_offsets, int E,
    int max_tiles, hipStream_t stream) {
 hipLaunchKernelGGL(( build_moe_tiles_kernel), dim3(1), dim3(64), 0, stream, tile_expert, tile_row0,
                                               seg_offsets, E, max_tiles);
}

extern "C" void launch_moe_gate_up(
    void* h_buf, const void* x, const void* w_gu, const int* tile_expert,
    const int* tile_row0, const int64_t* perm, const int* seg_offsets, int E,
    int topk, int H, int I, int max_tiles, bool gelu, float limit,
    hipStream_t stream) {
  dim3 grid(max_tiles, ceil_div(I, 64), 1);
  if (gelu)
   hipLaunchKernelGGL(( moe_gate_up_kernel<true>), dim3(grid), dim3(MOE_THREADS), 0, stream, 
        (uint16_t*)h_buf, (const uint16_t*)x, (const uint16_t*)w_gu,
        tile_expert, tile_row0, perm, seg_offsets, E, topk, H, I, limit);
  else
   hipLaunchKernelGGL(( moe_gate_up_kernel<false>), dim3(grid), dim3(MOE_THREADS), 0, stream, 
        (uint16_t*)h_buf, (const uint16_t*)x, (const uint16_t*)w_gu,
        tile_expert, tile_row0, perm, seg_offsets, E, topk, H, I, limit);
}

extern "C" void launch_moe_down(
    void* out, const void* h_buf, const void* w_down, const float* route_w,
    const int* tile_expert, const int* tile_row0, const int64_t* perm,
    const int* seg_offsets, int E, int topk, int H, int I, int max_tiles,
    hipStream_t stream) {
  dim3 grid(max_tiles, ceil_div(H, 256), 1);
 hipLaunchKernelGGL(( moe_down_kernel), dim3(grid), dim3(MOE_THREADS), 0, stream, 
      (float*)out, (const uint16_t*)h_buf, (const uint16_t*)w_down, route_w,
      tile_expert, tile_row0, perm, seg_offsets, E, topk, H, I);
}
