// Column sum for bias gradients: out[c] = sum_r dy[r][c], fp32 accum.
//
// (The initial round-2 reading that torch's reduce was "~50x the
// traffic bound / ~15 ms per step" was a profiler misattribution — a
// sloppy kernel-name match pulled in unrelated reduce_kernel launches.
// Measured properly, torch's bf16 dim-0 reduce is competitive at these
// shapes; this kernel only serves the off-by-default
// PROGEN_OVERLAP_WGRAD side-stream path.) Each thread owns 8 consecutive
// columns (one bf16x8 load per row), blocks stride the rows, partials
// combine with fp32 atomics (gridDim.y partials per column).

#include "common.h"

#define CS_BLOCK 256
#define CS_COLS (CS_BLOCK * 8)

__global__ __launch_bounds__(CS_BLOCK) void colsum_kernel(
    const short* __restrict__ dy, float* __restrict__ out, long long R,
    int C) {
  const int col0 = blockIdx.x * CS_COLS + (int)threadIdx.x * 8;
  if (col0 >= C) return;
  float acc[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) acc[j] = 0.f;
  for (long long r = blockIdx.y; r < R; r += gridDim.y) {
    bf16x8 v = *(const bf16x8*)(dy + r * C + col0);
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[j] += bf2f(((short*)&v)[j]);
  }
#pragma unroll
  for (int j = 0; j < 8; ++j) atomicAdd(out + col0 + j, acc[j]);
}

extern "C" {

void colsum_launch(const void* dy, float* out, long long R, int C,
                   hipStream_t stream) {
  const int gx = (C + CS_COLS - 1) / CS_COLS;
  // >> 256 blocks to fill the chip (the first cut capped at 256 = one
  // 4-wave block per CU and ran latency-bound, SLOWER than torch's
  // reduce); gy row-slices bound the fp32 atomics at gy per column
  int gy = 4096 / gx;
  if (gy < 64) gy = 64;
  if (gy > 1024) gy = 1024;
  if (gy > (int)R) gy = (int)R;
  dim3 grid(gx, gy);
  colsum_kernel<<<grid, CS_BLOCK, 0, stream>>>((const short*)dy, out, R, C);
}

}  // extern "C"
