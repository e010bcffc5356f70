// GLU-GELU epilogue of the feedforward (reference: progen.py:139-143):
//   glu:  y = a * gelu(g)  with (a, g) = split(h, 2, dim=-1)
//   gelu: y = gelu(h)
// Memory-bound elementwise kernels, 16 B/lane vectorized, grid-stride.

#include "common.h"

#define GLU_BLOCK 256

// 2-D grid (columns x row-stripes): the flat-index form cost two
// 64-bit integer divisions per iteration — a long VALU sequence on a
// kernel this loop-dense
template <typename VEC, bool IS_BF16>
__global__ __launch_bounds__(GLU_BLOCK) void glu_fwd_kernel(
    const VEC* __restrict__ h, VEC* __restrict__ y, long long rows, int Hv) {
  constexpr int VLEN = IS_BF16 ? 8 : 4;
  const int i = blockIdx.x * GLU_BLOCK + (int)threadIdx.x;
  if (i >= Hv) return;
  for (long long row = blockIdx.y; row < rows; row += gridDim.y) {
    VEC va = h[row * 2 * Hv + i];
    VEC vg = h[row * 2 * Hv + Hv + i];
    VEC o;
#pragma unroll
    for (int j = 0; j < VLEN; ++j) {
      float a = IS_BF16 ? bf2f(((short*)&va)[j]) : ((float*)&va)[j];
      float g = IS_BF16 ? bf2f(((short*)&vg)[j]) : ((float*)&vg)[j];
      float r = a * gelu_tanh(g);
      if (IS_BF16) ((short*)&o)[j] = f2bf(r); else ((float*)&o)[j] = r;
    }
    y[row * Hv + i] = o;
  }
}

template <typename VEC, bool IS_BF16>
__global__ __launch_bounds__(GLU_BLOCK) void glu_bwd_kernel(
    const VEC* __restrict__ dy, const VEC* __restrict__ h,
    VEC* __restrict__ dh, long long rows, int Hv) {
  constexpr int VLEN = IS_BF16 ? 8 : 4;
  const int i = blockIdx.x * GLU_BLOCK + (int)threadIdx.x;
  if (i >= Hv) return;
  for (long long row = blockIdx.y; row < rows; row += gridDim.y) {
    VEC va = h[row * 2 * Hv + i];
    VEC vg = h[row * 2 * Hv + Hv + i];
    VEC vdy = dy[row * Hv + i];
    VEC da, dg;
#pragma unroll
    for (int j = 0; j < VLEN; ++j) {
      float a = IS_BF16 ? bf2f(((short*)&va)[j]) : ((float*)&va)[j];
      float g = IS_BF16 ? bf2f(((short*)&vg)[j]) : ((float*)&vg)[j];
      float d = IS_BF16 ? bf2f(((short*)&vdy)[j]) : ((float*)&vdy)[j];
      float gv, gg;
      gelu_tanh_both(g, &gv, &gg);
      float rda = d * gv;
      float rdg = d * a * gg;
      if (IS_BF16) { ((short*)&da)[j] = f2bf(rda); ((short*)&dg)[j] = f2bf(rdg); }
      else { ((float*)&da)[j] = rda; ((float*)&dg)[j] = rdg; }
    }
    dh[row * 2 * Hv + i] = da;
    dh[row * 2 * Hv + Hv + i] = dg;
  }
}

template <typename VEC, bool IS_BF16>
__global__ __launch_bounds__(GLU_BLOCK) void gelu_fwd_kernel(
    const VEC* __restrict__ h, VEC* __restrict__ y, long long total) {
  constexpr int VLEN = IS_BF16 ? 8 : 4;
  for (long long idx = blockIdx.x * (long long)GLU_BLOCK + threadIdx.x;
       idx < total; idx += (long long)gridDim.x * GLU_BLOCK) {
    VEC v = h[idx];
    VEC o;
#pragma unroll
    for (int j = 0; j < VLEN; ++j) {
      float g = IS_BF16 ? bf2f(((short*)&v)[j]) : ((float*)&v)[j];
      float r = gelu_tanh(g);
      if (IS_BF16) ((short*)&o)[j] = f2bf(r); else ((float*)&o)[j] = r;
    }
    y[idx] = o;
  }
}

template <typename VEC, bool IS_BF16>
__global__ __launch_bounds__(GLU_BLOCK) void gelu_bwd_kernel(
    const VEC* __restrict__ dy, const VEC* __restrict__ h,
    VEC* __restrict__ dh, long long total) {
  constexpr int VLEN = IS_BF16 ? 8 : 4;
  for (long long idx = blockIdx.x * (long long)GLU_BLOCK + threadIdx.x;
       idx < total; idx += (long long)gridDim.x * GLU_BLOCK) {
    VEC v = h[idx];
    VEC vdy = dy[idx];
    VEC o;
#pragma unroll
    for (int j = 0; j < VLEN; ++j) {
      float g = IS_BF16 ? bf2f(((short*)&v)[j]) : ((float*)&v)[j];
      float d = IS_BF16 ? bf2f(((short*)&vdy)[j]) : ((float*)&vdy)[j];
      float r = d * gelu_tanh_grad(g);
      if (IS_BF16) ((short*)&o)[j] = f2bf(r); else ((float*)&o)[j] = r;
    }
    dh[idx] = o;
  }
}

static inline int glu_grid(long long total) {
  long long g = (total + GLU_BLOCK - 1) / GLU_BLOCK;
  if (g > 2048) g = 2048;  // grid-stride the rest (G11)
  return (int)g;
}

// 2-D grid for the split (glu) kernels: x covers the Hv columns, y
// row-stripes sized so x*y lands near 4096 blocks
static inline dim3 glu_grid2(long long rows, int Hv) {
  int gx = (Hv + GLU_BLOCK - 1) / GLU_BLOCK;
  long long gy = 4096 / gx;
  if (gy < 1) gy = 1;
  if (gy > rows) gy = rows;
  return dim3(gx, (unsigned)gy);
}

extern "C" {

void glu_fwd_launch(const void* h, void* y, long long rows, int H,
                    bool is_bf16, hipStream_t stream) {
  if (is_bf16) {
    int Hv = H / 8;
    glu_fwd_kernel<bf16x8, true><<<glu_grid2(rows, Hv), GLU_BLOCK, 0, stream>>>(
        (const bf16x8*)h, (bf16x8*)y, rows, Hv);
  } else {
    int Hv = H / 4;
    glu_fwd_kernel<f32x4, false><<<glu_grid2(rows, Hv), GLU_BLOCK, 0, stream>>>(
        (const f32x4*)h, (f32x4*)y, rows, Hv);
  }
}

void glu_bwd_launch(const void* dy, const void* h, void* dh, long long rows,
                    int H, bool is_bf16, hipStream_t stream) {
  if (is_bf16) {
    int Hv = H / 8;
    glu_bwd_kernel<bf16x8, true><<<glu_grid2(rows, Hv), GLU_BLOCK, 0, stream>>>(
        (const bf16x8*)dy, (const bf16x8*)h, (bf16x8*)dh, rows, Hv);
  } else {
    int Hv = H / 4;
    glu_bwd_kernel<f32x4, false><<<glu_grid2(rows, Hv), GLU_BLOCK, 0, stream>>>(
        (const f32x4*)dy, (const f32x4*)h, (f32x4*)dh, rows, Hv);
  }
}

void gelu_fwd_launch(const void* h, void* y, long long total_elems,
                     bool is_bf16, hipStream_t stream) {
  long long tv = total_elems / (is_bf16 ? 8 : 4);
  if (is_bf16)
    gelu_fwd_kernel<bf16x8, true><<<glu_grid(tv), GLU_BLOCK, 0, stream>>>(
        (const bf16x8*)h, (bf16x8*)y, tv);
  else
    gelu_fwd_kernel<f32x4, false><<<glu_grid(tv), GLU_BLOCK, 0, stream>>>(
        (const f32x4*)h, (f32x4*)y, tv);
}

void gelu_bwd_launch(const void* dy, const void* h, void* dh,
                     long long total_elems, bool is_bf16, hipStream_t stream) {
  long long tv = total_elems / (is_bf16 ? 8 : 4);
  if (is_bf16)
    gelu_bwd_kernel<bf16x8, true><<<glu_grid(tv), GLU_BLOCK, 0, stream>>>(
        (const bf16x8*)dy, (const bf16x8*)h, (bf16x8*)dh, tv);
  else
    gelu_bwd_kernel<f32x4, false><<<glu_grid(tv), GLU_BLOCK, 0, stream>>>(
        (const f32x4*)dy, (const f32x4*)h, (f32x4*)dh, tv);
}

}  // extern "C"
