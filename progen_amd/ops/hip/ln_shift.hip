// Fused scale-only LayerNorm + token shift, forward and backward.
//
// Implements the LN->shift prologue of both branches
// (reference: progen.py:22,43-46,74-77,132-135) as one memory-bound pass:
// the block normalizes row (b, n) and writes the first ceil(D/2) channels
// into row n+1 (the token shift) and the rest into row n, so the shifted
// activation never exists as a separate tensor.
//
// Memory-bound: vectorized 16 B/lane loads (bf16x8), fp32 statistics,
// one block per row forward; grid-strided rows backward with per-block
// dweight partials reduced on the host side.

#include "common.h"

#define LN_BLOCK 256
#define LN_WAVES (LN_BLOCK / WAVE)

// block-reduce two per-thread partial sums through a LN_WAVES-float
// shared scratch (all threads must reach this together)
__device__ __forceinline__ void ln_block_reduce2(float& a, float& b,
                                                 float* red, int wid,
                                                 int lane) {
  a = wave_sum(a);
  b = wave_sum(b);
  if (lane == 0) red[wid] = a;
  __syncthreads();
  if (wid == 0) {
    float t = (lane < LN_WAVES) ? red[lane] : 0.f;
#pragma unroll
    for (int off = LN_WAVES / 2; off > 0; off >>= 1) t += __shfl_xor(t, off, 64);
    if (lane == 0) red[0] = t;
  }
  __syncthreads();
  a = red[0];
  __syncthreads();
  if (lane == 0) red[wid] = b;
  __syncthreads();
  if (wid == 0) {
    float t = (lane < LN_WAVES) ? red[lane] : 0.f;
#pragma unroll
    for (int off = LN_WAVES / 2; off > 0; off >>= 1) t += __shfl_xor(t, off, 64);
    if (lane == 0) red[0] = t;
  }
  __syncthreads();
  b = red[0];
  __syncthreads();
}

// ---------------------------------------------------------------------------
// forward
// ---------------------------------------------------------------------------

// RES: fuse the residual add s = x + res (bf16 rounding identical to an
// eager bf16 add); stats and y are computed on s, and s is written out as
// the new residual stream — the separate elementwise add (and its
// backward-side grad accumulation) disappears.
template <typename VEC, bool SHIFT, bool IS_BF16, bool RES>
__global__ __launch_bounds__(LN_BLOCK) void ln_shift_fwd_kernel(
    const VEC* __restrict__ x, const VEC* __restrict__ res,
    const VEC* __restrict__ g, VEC* __restrict__ y, VEC* __restrict__ s_out,
    float* __restrict__ mean, float* __restrict__ rstd, int N, int Dv,
    float eps) {
  // Dv = D / VLEN (vector units); each VEC is 8 bf16 or 4 f32 (16 B)
  constexpr int VLEN = IS_BF16 ? 8 : 4;
  const int row = blockIdx.x;
  const int n = row % N;
  const long long base = (long long)row * Dv;

  __shared__ float red[LN_WAVES];
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int D = Dv * VLEN;
  const int halfv = (D / 2) / VLEN;  // D even, half % VLEN == 0 (host-checked)

  if (Dv <= LN_BLOCK) {
    // fast path (every production D): each thread owns <= 1 vector, so
    // the row stays in REGISTERS between the stats pass and the
    // normalize pass — no second global read
    const int i = threadIdx.x;
    const bool act = i < Dv;
    VEC v;
    float s = 0.f, ss = 0.f;
    if (act) {
      v = x[base + i];
      if (RES) {
        VEC rv = res[base + i];
        VEC sv;
#pragma unroll
        for (int j = 0; j < VLEN; ++j) {
          if (IS_BF16) {
            ((short*)&sv)[j] = f2bf(bf2f(((short*)&v)[j]) + bf2f(((short*)&rv)[j]));
          } else {
            ((float*)&sv)[j] = ((float*)&v)[j] + ((float*)&rv)[j];
          }
        }
        s_out[base + i] = sv;
        v = sv;
      }
#pragma unroll
      for (int j = 0; j < VLEN; ++j) {
        float f = IS_BF16 ? bf2f(((short*)&v)[j]) : ((float*)&v)[j];
        s += f;
        ss += f * f;
      }
    }
    ln_block_reduce2(s, ss, red, wid, lane);
    const float mu = s / D;
    const float var = fmaxf(ss / D - mu * mu, 0.f);
    const float rs = rsqrtf(var + eps);
    if (threadIdx.x == 0) {
      mean[row] = mu;
      rstd[row] = rs;
    }
    if (act) {
      VEC gw = g[i];
      VEC o;
#pragma unroll
      for (int j = 0; j < VLEN; ++j) {
        float f = IS_BF16 ? bf2f(((short*)&v)[j]) : ((float*)&v)[j];
        float gj = IS_BF16 ? bf2f(((short*)&gw)[j]) : ((float*)&gw)[j];
        float r = (f - mu) * rs * gj;
        if (IS_BF16) ((short*)&o)[j] = f2bf(r);
        else ((float*)&o)[j] = r;
      }
      if (!SHIFT || i >= halfv) {
        y[base + i] = o;
      } else if (n + 1 < N) {
        y[base + Dv + i] = o;
      }
    }
    if (SHIFT && n == 0) {
      VEC z;
#pragma unroll
      for (int j = 0; j < VLEN; ++j) {
        if (IS_BF16) ((short*)&z)[j] = 0; else ((float*)&z)[j] = 0.f;
      }
      for (int k = threadIdx.x; k < halfv; k += LN_BLOCK) y[base + k] = z;
    }
    return;
  }

  float s = 0.f, ss = 0.f;
  for (int i = threadIdx.x; i < Dv; i += LN_BLOCK) {
    VEC v = x[base + i];
    if (RES) {
      VEC rv = res[base + i];
      VEC sv;
#pragma unroll
      for (int j = 0; j < VLEN; ++j) {
        if (IS_BF16) {
          ((short*)&sv)[j] = f2bf(bf2f(((short*)&v)[j]) + bf2f(((short*)&rv)[j]));
        } else {
          ((float*)&sv)[j] = ((float*)&v)[j] + ((float*)&rv)[j];
        }
      }
      s_out[base + i] = sv;
      v = sv;
    }
#pragma unroll
    for (int j = 0; j < VLEN; ++j) {
      float f = IS_BF16 ? bf2f(((short*)&v)[j]) : ((float*)&v)[j];
      s += f;
      ss += f * f;
    }
  }
  ln_block_reduce2(s, ss, red, wid, lane);

  const float mu = s / D;
  const float var = fmaxf(ss / D - mu * mu, 0.f);
  const float rs = rsqrtf(var + eps);
  if (threadIdx.x == 0) {
    mean[row] = mu;
    rstd[row] = rs;
  }

  for (int i = threadIdx.x; i < Dv; i += LN_BLOCK) {
    // with RES the summed row was just written: the re-read hits L1/L2
    VEC v = RES ? s_out[base + i] : x[base + i];
    VEC gw = g[i];
    VEC o;
#pragma unroll
    for (int j = 0; j < VLEN; ++j) {
      float f = IS_BF16 ? bf2f(((short*)&v)[j]) : ((float*)&v)[j];
      float gj = IS_BF16 ? bf2f(((short*)&gw)[j]) : ((float*)&gw)[j];
      float r = (f - mu) * rs * gj;
      if (IS_BF16) ((short*)&o)[j] = f2bf(r);
      else ((float*)&o)[j] = r;
    }
    if (!SHIFT || i >= halfv) {
      y[base + i] = o;
    } else if (n + 1 < N) {  // shift half -> next row (within the sequence)
      y[base + Dv + i] = o;
    }
  }
  if (SHIFT && n == 0) {  // first row's shift half is the zero pad
    VEC z;
#pragma unroll
    for (int j = 0; j < VLEN; ++j) {
      if (IS_BF16) ((short*)&z)[j] = 0; else ((float*)&z)[j] = 0.f;
    }
    for (int i = threadIdx.x; i < halfv; i += LN_BLOCK) y[base + i] = z;
  }
}

// ---------------------------------------------------------------------------
// backward
// ---------------------------------------------------------------------------
// e[d] = upstream grad routed back through the shift:
//   e[d < half] = dy[row+1][d] (0 at the last row of a sequence)
//   e[d >= half] = dy[row][d]
// dx = rs * (e*g - mean(e*g) - xhat * mean(e*g*xhat));  dw += e * xhat

// DS: also add the gradient flowing into the summed stream s (the fused
// variant returns s as a second output; dx then serves as the gradient of
// BOTH addends, since d(x + res) fans out identically).
template <typename VEC, bool SHIFT, bool IS_BF16, bool DS>
__global__ __launch_bounds__(LN_BLOCK) void ln_shift_bwd_kernel(
    const VEC* __restrict__ dy, const VEC* __restrict__ ds,
    const VEC* __restrict__ x,
    const VEC* __restrict__ g, const float* __restrict__ mean,
    const float* __restrict__ rstd, VEC* __restrict__ dx,
    float* __restrict__ dw_part, int R, int N, int Dv) {
  constexpr int VLEN = IS_BF16 ? 8 : 4;
  const int D = Dv * VLEN;
  const int halfv = (D / 2) / VLEN;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* dw_lds = (float*)smem;             // D floats
  float* red = (float*)(smem + (size_t)D * 4);  // LN_WAVES floats

  for (int i = threadIdx.x; i < D; i += LN_BLOCK) dw_lds[i] = 0.f;
  __syncthreads();

  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;

  for (int row = blockIdx.x; row < R; row += gridDim.x) {
    const int n = row % N;
    const long long base = (long long)row * Dv;
    const float mu = mean[row];
    const float rs = rstd[row];
    const float inv_d = 1.0f / D;

    if (Dv <= LN_BLOCK) {
      // fast path: each thread owns <= 1 vector; e/x/g stay in
      // registers between the two phases — no second global read
      const int i = threadIdx.x;
      const bool act = i < Dv;
      VEC e, v, gw, dsv;
      bool have_e = false;
      float a = 0.f, bsum = 0.f;
      if (act) {
        const bool shifted = SHIFT && (i < halfv);
        have_e = true;
        if (shifted) {
          if (n + 1 < N) e = dy[base + Dv + i];
          else have_e = false;
        } else {
          e = dy[base + i];
        }
        v = x[base + i];
        gw = g[i];
        if (DS) dsv = ds[base + i];
#pragma unroll
        for (int j = 0; j < VLEN; ++j) {
          float ej = have_e ? (IS_BF16 ? bf2f(((short*)&e)[j]) : ((float*)&e)[j]) : 0.f;
          float f = IS_BF16 ? bf2f(((short*)&v)[j]) : ((float*)&v)[j];
          float gj = IS_BF16 ? bf2f(((short*)&gw)[j]) : ((float*)&gw)[j];
          float xh = (f - mu) * rs;
          a += ej * gj * xh;
          bsum += ej * gj;
        }
      }
      ln_block_reduce2(a, bsum, red, wid, lane);
      if (act) {
        VEC o;
#pragma unroll
        for (int j = 0; j < VLEN; ++j) {
          float ej = have_e ? (IS_BF16 ? bf2f(((short*)&e)[j]) : ((float*)&e)[j]) : 0.f;
          float f = IS_BF16 ? bf2f(((short*)&v)[j]) : ((float*)&v)[j];
          float gj = IS_BF16 ? bf2f(((short*)&gw)[j]) : ((float*)&gw)[j];
          float xh = (f - mu) * rs;
          float dxv = rs * (ej * gj - bsum * inv_d - xh * a * inv_d);
          if (DS) dxv += IS_BF16 ? bf2f(((short*)&dsv)[j]) : ((float*)&dsv)[j];
          if (IS_BF16) ((short*)&o)[j] = f2bf(dxv);
          else ((float*)&o)[j] = dxv;
          dw_lds[i * VLEN + j] += ej * xh;  // thread-private index: no race
        }
        dx[base + i] = o;
      }
      __syncthreads();
      continue;
    }

    float a = 0.f, bsum = 0.f;  // a = sum(e*g*xhat), bsum = sum(e*g)
    for (int i = threadIdx.x; i < Dv; i += LN_BLOCK) {
      const bool shifted = SHIFT && (i < halfv);
      VEC e;
      bool have_e = true;
      if (shifted) {
        if (n + 1 < N) e = dy[base + Dv + i];
        else have_e = false;
      } else {
        e = dy[base + i];
      }
      VEC v = x[base + i];
      VEC gw = g[i];
#pragma unroll
      for (int j = 0; j < VLEN; ++j) {
        float ej = have_e ? (IS_BF16 ? bf2f(((short*)&e)[j]) : ((float*)&e)[j]) : 0.f;
        float f = IS_BF16 ? bf2f(((short*)&v)[j]) : ((float*)&v)[j];
        float gj = IS_BF16 ? bf2f(((short*)&gw)[j]) : ((float*)&gw)[j];
        float xh = (f - mu) * rs;
        a += ej * gj * xh;
        bsum += ej * gj;
      }
    }
    ln_block_reduce2(a, bsum, red, wid, lane);

    for (int i = threadIdx.x; i < Dv; i += LN_BLOCK) {
      const bool shifted = SHIFT && (i < halfv);
      VEC e;
      bool have_e = true;
      if (shifted) {
        if (n + 1 < N) e = dy[base + Dv + i];
        else have_e = false;
      } else {
        e = dy[base + i];
      }
      VEC v = x[base + i];
      VEC gw = g[i];
      VEC dsv;
      if (DS) dsv = ds[base + i];
      VEC o;
#pragma unroll
      for (int j = 0; j < VLEN; ++j) {
        float ej = have_e ? (IS_BF16 ? bf2f(((short*)&e)[j]) : ((float*)&e)[j]) : 0.f;
        float f = IS_BF16 ? bf2f(((short*)&v)[j]) : ((float*)&v)[j];
        float gj = IS_BF16 ? bf2f(((short*)&gw)[j]) : ((float*)&gw)[j];
        float xh = (f - mu) * rs;
        float dxv = rs * (ej * gj - bsum * inv_d - xh * a * inv_d);
        if (DS) dxv += IS_BF16 ? bf2f(((short*)&dsv)[j]) : ((float*)&dsv)[j];
        if (IS_BF16) ((short*)&o)[j] = f2bf(dxv);
        else ((float*)&o)[j] = dxv;
        dw_lds[i * VLEN + j] += ej * xh;  // thread-private index: no race
      }
      dx[base + i] = o;
    }
    __syncthreads();
  }

  float* out = dw_part + (size_t)blockIdx.x * D;
  for (int i = threadIdx.x; i < D; i += LN_BLOCK) out[i] = dw_lds[i];
}

// ---------------------------------------------------------------------------
// launchers
// ---------------------------------------------------------------------------

extern "C" {

void ln_shift_fwd_launch(const void* x, const void* res, const void* g,
                         void* y, void* s_out, float* mean, float* rstd,
                         int R, int N, int D, float eps, bool shift,
                         bool is_bf16, hipStream_t stream) {
  dim3 grid(R), block(LN_BLOCK);
#define LNF(VEC, SH, BF, RS)                                              \
  ln_shift_fwd_kernel<VEC, SH, BF, RS><<<grid, block, 0, stream>>>(       \
      (const VEC*)x, (const VEC*)res, (const VEC*)g, (VEC*)y,             \
      (VEC*)s_out, mean, rstd, N, Dv, eps)
  if (is_bf16) {
    int Dv = D / 8;
    if (shift) { if (res) LNF(bf16x8, true, true, true); else LNF(bf16x8, true, true, false); }
    else       { if (res) LNF(bf16x8, false, true, true); else LNF(bf16x8, false, true, false); }
  } else {
    int Dv = D / 4;
    if (shift) { if (res) LNF(f32x4, true, false, true); else LNF(f32x4, true, false, false); }
    else       { if (res) LNF(f32x4, false, false, true); else LNF(f32x4, false, false, false); }
  }
#undef LNF
}

void ln_shift_bwd_launch(const void* dy, const void* ds, const void* x,
                         const void* g, const float* mean, const float* rstd,
                         void* dx, float* dw_part, int nblocks, int R, int N,
                         int D, bool shift, bool is_bf16, hipStream_t stream) {
  dim3 grid(nblocks), block(LN_BLOCK);
  size_t lds = (size_t)D * 4 + LN_WAVES * 4 + 16;
#define LNB(VEC, SH, BF, WDS)                                             \
  ln_shift_bwd_kernel<VEC, SH, BF, WDS><<<grid, block, lds, stream>>>(    \
      (const VEC*)dy, (const VEC*)ds, (const VEC*)x, (const VEC*)g, mean, \
      rstd, (VEC*)dx, dw_part, R, N, Dv)
  if (is_bf16) {
    int Dv = D / 8;
    if (shift) { if (ds) LNB(bf16x8, true, true, true); else LNB(bf16x8, true, true, false); }
    else       { if (ds) LNB(bf16x8, false, true, true); else LNB(bf16x8, false, true, false); }
  } else {
    int Dv = D / 4;
    if (shift) { if (ds) LNB(f32x4, true, false, true); else LNB(f32x4, true, false, false); }
    else       { if (ds) LNB(f32x4, false, false, true); else LNB(f32x4, false, false, false); }
  }
#undef LNB
}

}  // extern "C"
