#include "hip/hip_runtime.h"
// Skinny-M dense GEMM: C[M,N] = X[M,K] . W[N,K]^T (+bias), M <= 256.
//
// Decode-step GEMMs are M = batch (1-256) against multi-hundred-MB weight
// matrices — pure weight streaming. hipBLASLt's tile picks run 2-5x off the
// HBM roofline at these shapes (measured, profiles/README.md). Structure:
// one workgroup per 64-column W slab; the X K-slab (M x 128) is staged in LDS
// once per iteration (cooperative coalesced loads, padded rows against bank
// conflicts) and every wave's MFMA A-fragments come from LDS; each W row is
// streamed exactly once per slab by the lane that owns that output column.

#include "common.h"

#define SG_THREADS 256
#define SG_BK 128           // K elements staged per iteration
#define SG_PROW (SG_BK + 8) // padded LDS row (elements)

template <int MTILES>
__global__ __launch_bounds__(SG_THREADS) void skinny_gemm_kernel(
    uint16_t* __restrict__ c,        // [M, N] bf16 rows at c_stride
    const uint16_t* __restrict__ x,  // [M, K] rows at x_stride
    const uint16_t* __restrict__ w,  // [N, K]
    const uint16_t* __restrict__ bias,  // [N] or nullptr
    const int M, const int N, const int K, const int64_t x_stride,
    const int64_t c_stride) {
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int l15 = lane & 15;
  const int l4 = lane >> 4;

  const int n = blockIdx.x * 64 + wid * 16 + l15;  // this lane's W row / C col
  const uint16_t* wrow = w + (size_t)n * K;

  __shared__ uint16_t Xl[MTILES * 16 * SG_PROW];

  f32x4v acc[MTILES] = {};
  for (int k0 = 0; k0 < K; k0 += SG_BK) {
    __syncthreads();
    // stage X[:, k0:k0+128]: (row, 16B chunk) pairs over all threads so small
    // M still engages the whole workgroup, chunks coalesce within a row.
    // Rows past M are zero-filled (read by padded MFMA rows).
    for (int idx = tid; idx < MTILES * 16 * (SG_BK / 8); idx += SG_THREADS) {
      const int row = idx / (SG_BK / 8);
      const int cc = idx % (SG_BK / 8);
      int4 val = make_int4(0, 0, 0, 0);
      if (row < M)
        val = *reinterpret_cast<const int4*>(
            x + (size_t)row * x_stride + k0 + cc * 8);
      *reinterpret_cast<int4*>(Xl + row * SG_PROW + cc * 8) = val;
    }
    __syncthreads();

#pragma unroll
    for (int ks = 0; ks < SG_BK / 32; ++ks) {
      const bf16x8v bfrag = *reinterpret_cast<const bf16x8v*>(
          wrow + k0 + ks * 32 + l4 * 8);
#pragma unroll
      for (int mt = 0; mt < MTILES; ++mt) {
        const int row = mt * 16 + l15;
        // stale LDS rows past M contribute to padded acc rows only, which are
        // never stored — but keep them zeroed for NaN safety
        const bf16x8v afrag = *reinterpret_cast<const bf16x8v*>(
            Xl + row * SG_PROW + ks * 32 + l4 * 8);
        acc[mt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag,
                                                          acc[mt], 0, 0, 0);
      }
    }
  }

  const float b = bias != nullptr ? bf16_bits_to_f32(bias[n]) : 0.f;
#pragma unroll
  for (int mt = 0; mt < MTILES; ++mt) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = mt * 16 + l4 * 4 + r;
      if (row < M)
        c[(size_t)row * c_stride + n] = f32_to_bf16_bits(acc[mt][r] + b);
    }
  }
}

extern "C" void launch_skinny_gemm(
    void* c, const void* x, const void* w, const void* bias, int M, int N,
    int K, int64_t x_stride, int64_t c_stride, hipStream_t stream,
    bool* launched) {
  *launched = false;
  if (M > 256 || N % 64 != 0 || K % SG_BK != 0) return;
  dim3 grid(N / 64, 1, 1);
#define SG_LAUNCH(MT)                                                      \
 hipLaunchKernelGGL(( skinny_gemm_kernel<MT>), dim3(grid), dim3(SG_THREADS), 0, stream,                  \
      (uint16_t*)c, (const uint16_t*)x, (const uint16_t*)w,                \
      (const uint16_t*)bias, M, N, K, x_stride, c_stride);                 \
  *launched = true;
  if (M <= 16) { SG_LAUNCH(1) }
  else if (M <= 32) { SG_LAUNCH(2) }
  else if (M <= 64) { SG_LAUNCH(4) }
  else if (M <= 128) { SG_LAUNCH(8) }
  else { SG_LAUNCH(16) }
#undef SG_LAUNCH
}
