// K6: gradient accumulation — sum of G per-worker gradient vectors
// (G x D f32, row-major) -> D f32 (APRIL-ANN's axpy loop,
// common.lua:127-136).
//
// Two implementations, A/B-measured (profiles/gradsum_ab.log):
//   colsum_f32_valu — float4-vectorized column sum, grid-stride; the
//     operation reads G*D*4 bytes once and writes D*4: pure HBM-bound,
//     so this is the roofline path.
//   colsum_f32_mfma — mfma_f32_16x16x4_f32 with an all-ones A operand:
//     C[i][j] = sum_k B[k][j], i.e. 16 column-sums per MFMA with exact
//     f32 numerics (the f32-in MFMA is a k-ordered fmaf chain — guide
//     §3).  Kept for the record: a rank-1 reduction cannot beat the
//     bandwidth bound, and its 4-byte/lane loads under-utilize the bus;
//     measured slower than the VALU path (see profiles/).

#include "common.h"

typedef float f32;

__global__ __launch_bounds__(256) void colsum_f32_valu_kernel(
    const f32* __restrict__ grads, long G, long D, f32* __restrict__ out) {
  long c4 = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 4;
  long stride = (long)gridDim.x * blockDim.x * 4;
  for (; c4 + 3 < D; c4 += stride) {
    float4 acc = {0.f, 0.f, 0.f, 0.f};
    for (long g = 0; g < G; ++g) {
      const float4 v = *(const float4*)&grads[g * D + c4];
      acc.x += v.x;
      acc.y += v.y;
      acc.z += v.z;
      acc.w += v.w;
    }
    *(float4*)&out[c4] = acc;
  }
  // tail columns (D not a multiple of 4): first wave handles them
  if (blockIdx.x == 0 && threadIdx.x < (D & 3)) {
    long c = (D & ~3L) + threadIdx.x;
    f32 acc = 0.f;
    for (long g = 0; g < G; ++g) acc += grads[g * D + c];
    out[c] = acc;
  }
}

typedef __attribute__((ext_vector_type(4))) float f32x4;

__global__ __launch_bounds__(64) void colsum_f32_mfma_kernel(
    const f32* __restrict__ grads, long G, long D, f32* __restrict__ out) {
  // one wave per block; each wave owns 16-column tiles, grid-strided
  int lane = threadIdx.x;
  long tile = blockIdx.x;
  long ntiles = (D + 15) / 16;
  for (; tile < ntiles; tile += gridDim.x) {
    long c0 = tile * 16;
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    long col = c0 + (lane & 15);
    bool incol = col < D;
    for (long g0 = 0; g0 < G; g0 += 4) {
      long row = g0 + (lane >> 4);
      f32 b = (incol && row < G) ? grads[row * D + col] : 0.f;
      // A = 1.0 everywhere: C[i][j] = sum_k B[k][j]
      acc = __builtin_amdgcn_mfma_f32_16x16x4f32(1.0f, b, acc, 0, 0, 0);
    }
    // C row 0 (lanes 0-15, reg 0) holds the column sums (C/D map:
    // col = lane&15, row = (lane>>4)*4 + reg — guide §3)
    if (lane < 16 && incol) out[col] = acc[0];
  }
}
