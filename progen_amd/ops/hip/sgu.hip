// SGU — gMLP spatial gating unit kernels (reference: progen.py:151-185).
//
//   gate_out[b,m,d] = sum_{k<=m} W[m,k] * g_ln[b,k,d] + bias[m]
//   out[b,m,d]      = xa[b,m,d] * gate_out[b,m,d]
//
// The causal (tril) mask is baked into the K-tile iteration — tiles
// above the diagonal are skipped, the diagonal tile is masked at W
// staging — instead of materializing the masked (n, n) matrix the
// reference multiplies through XLA (progen.py:179-181).
//
// Three kernels:
//   sgu_fwd_kernel    : the causal spatial matmul + bias + gate multiply
//                       (saves gate_out for backward)
//   sgu_dgate_kernel  : dg_ln[k,d] = sum_{m>=k} W[m,k] * t[m,d]
//                       (t = dy * xa, computed on the torch side)
//   sgu_dw_kernel     : dW[m,k] = sum_{b,d} t[b,m,d] * g_ln[b,k,d], k<=m
//                       (lower-triangle tiles only, split-K over (b,d)
//                       with fp32 atomicAdd combine)
//
// All on mfma_f32_16x16x32_bf16; W is staged through LDS (bf16,
// XOR-swizzled); the g_ln/t channel tiles use the same [d][k]
// scatter-transposed LDS image as the attention V tile. sgu_dw reads
// both operands fragment-shaped straight from global (both layouts are
// naturally k-contiguous; the W tile reuse across (b, d) keeps them
// L2-resident).

#include "common.h"

#define SGU_WAVES 4
#define SGU_BLOCK (SGU_WAVES * WAVE)

// XOR windows for tr-read images (derivations: wgrad probe header for
// 512-B rows, attention_bwd.hip for 128-B rows)
__device__ __forceinline__ int uk512(int k) { return (k & 3) | ((k & 8) >> 1); }
__device__ __forceinline__ int uk128(int k) { return ((k & 2) >> 1) | ((k & 8) >> 2); }

typedef __attribute__((__vector_size__(4 * sizeof(__bf16)))) __bf16 sgu_bf16x4t;
#define SGU_AS3 __attribute__((address_space(3)))

// one MFMA fragment from two ds_read_b64_tr_b16 of a row-major image
__device__ __forceinline__ bf16x8 sgu_frag_tr(const char* img, int rowbytes,
                                              int r1, int r2, int colb1,
                                              int colb2) {
  auto p1 = (SGU_AS3 sgu_bf16x4t*)(img + r1 * rowbytes + colb1);
  auto p2 = (SGU_AS3 sgu_bf16x4t*)(img + r2 * rowbytes + colb2);
  sgu_bf16x4t a = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p1);
  sgu_bf16x4t b = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p2);
  bf16x8 o;
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    ((__bf16*)&o)[j] = a[j];
    ((__bf16*)&o)[j + 4] = b[j];
  }
  return o;
}

__device__ __forceinline__ int swz(int row, int byte_in_row) {
  return (byte_in_row ^ ((row & 7) << 4));
}

// ---------------------------------------------------------------------------
// forward: block = 256 m-rows (4 waves x 64) x 64 channels
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(SGU_BLOCK) void sgu_fwd_kernel(
    const short* __restrict__ xa,    // (B, N, D) bf16
    const short* __restrict__ g_ln,  // (B, N, D) bf16 (LN'd gate half)
    const short* __restrict__ w,     // (n, n) bf16 spatial weights
    const float* __restrict__ bias,  // (n,) fp32
    short* __restrict__ out,         // (B, N, D) bf16
    short* __restrict__ gate_out,    // (B, N, D) bf16 (saved for bwd)
    int B, int N, int D) {
  const int mblk = blockIdx.x;   // 256-row m block
  const int dblk = blockIdx.y;   // 64-channel block
  const int batch = blockIdx.z;

  const int lane = threadIdx.x % WAVE;
  const int wid = threadIdx.x / WAVE;
  const int l15 = lane & 15;
  const int l4 = lane >> 4;

  const int m0 = mblk * 256 + wid * 64;  // wave's 64 m rows
  const int d0 = dblk * 64;
  const long long bND = (long long)batch * N * D;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  // LINEAR [k=64][d=64] image (128-B rows), B-fragments by tr read —
  // replaces the 16-b16-scatter-per-unit transposed staging
  char* gt_lds = smem;  // 8 KiB

  f32x4 acc[4][4];
#pragma unroll
  for (int m = 0; m < 4; ++m)
#pragma unroll
    for (int n = 0; n < 4; ++n) acc[m][n] = (f32x4){0.f, 0.f, 0.f, 0.f};

  const int ktiles = (mblk * 256 + 255) / 64 + 1;  // causal bound for block
  const int my_kmax = (m0 + 63) / 64;              // wave's causal bound

  for (int t = 0; t < ktiles; ++t) {
    // cooperative stage: g_ln[k0..k0+63][d0..d0+63] -> gt_lds [d][k]
    __syncthreads();
    {
      const int flat = threadIdx.x;  // 512 units / 256 threads = 2 passes
#pragma unroll
      for (int pass = 0; pass < 2; ++pass) {
        const int u = pass * SGU_BLOCK + flat;
        const int k = u >> 3;
        const int dd = (u & 7) * 8;
        bf16x8 v = *(const bf16x8*)(g_ln + bND + (long long)(t * 64 + k) * D +
                                    d0 + dd);
        *(bf16x8*)(gt_lds + k * 128 + ((dd * 2) ^ (uk128(k) * 32))) = v;
      }
    }
    __syncthreads();

    if (t <= my_kmax) {
      const bool diag = (t * 64) > m0 - 64 && (t * 64) <= m0 + 63;  // overlaps
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        // B-fragments [d][k-contig] by tr read of the linear [k][d]
        // image, hoisted for reuse across the 4 A-fragments
        const int r1 = ks * 32 + l4 * 8 + (l15 >> 2);
        const int r2 = r1 + 4;
        bf16x8 bfr[4];
#pragma unroll
        for (int n = 0; n < 4; ++n) {
          const int cb = (n * 16 + (l15 & 3) * 4) * 2;
          bfr[n] = sgu_frag_tr(gt_lds, 128, r1, r2,
                               cb ^ (uk128(r1) * 32), cb ^ (uk128(r2) * 32));
        }
#pragma unroll
        for (int m = 0; m < 4; ++m) {
          // A-fragment: W[m0+m*16+l15][t*64 + ks*32 + 8*l4 .. +8] bf16
          const int row = m0 + m * 16 + l15;
          const int kk0 = t * 64 + ks * 32 + 8 * l4;
          bf16x8 af = *(const bf16x8*)(w + (long long)row * N + kk0);
          if (diag) {  // causal mask within the diagonal band
#pragma unroll
            for (int j = 0; j < 8; ++j)
              if (kk0 + j > row) ((short*)&af)[j] = 0;
          }
#pragma unroll
          for (int n = 0; n < 4; ++n)
            acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                af, bfr[n], acc[m][n], 0, 0, 0);
        }
      }
    }
  }

  // epilogue: + bias, save gate_out, * xa, store out
#pragma unroll
  for (int m = 0; m < 4; ++m)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = m0 + m * 16 + l4 * 4 + r;
      const float bv = bias[row];
#pragma unroll
      for (int n = 0; n < 4; ++n) {
        const int d = d0 + n * 16 + l15;
        const long long off = bND + (long long)row * D + d;
        float g = ((float*)&acc[m][n])[r] + bv;
        gate_out[off] = f2bf(g);
        out[off] = f2bf(bf2f(xa[off]) * g);
      }
    }
}

// ---------------------------------------------------------------------------
// dgate: dg[k,d] = sum_{m>=k} W[m,k] t[m,d]
// block = 256 k-rows (4 waves x 64) x 64 channels; iterate m-tiles
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(SGU_BLOCK) void sgu_dgate_kernel(
    const short* __restrict__ t_in,  // (B, N, D) bf16, t = dy * xa
    const short* __restrict__ w,     // (n, n) bf16
    short* __restrict__ dg,          // (B, N, D) bf16
    int B, int N, int D) {
  const int kblk = blockIdx.x;
  const int dblk = blockIdx.y;
  const int batch = blockIdx.z;

  const int lane = threadIdx.x % WAVE;
  const int wid = threadIdx.x / WAVE;
  const int l15 = lane & 15;
  const int l4 = lane >> 4;

  const int k0 = kblk * 256 + wid * 64;  // wave's 64 k rows
  const int d0 = dblk * 64;
  const long long bND = (long long)batch * N * D;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  // LINEAR images read with ds_read_b64_tr_b16 (the old scatter-
  // transposed staging was 64 b16 writes per thread per tile and ran
  // the kernel at 71 TF/s — the wgrad-gen-1 disease):
  char* tt_lds = smem;          // [m=64][d=64] linear, 128-B rows, 8 KiB
  char* wt_lds = smem + 8192;   // [m=64][k=256] linear, 512-B rows, 32 KiB

  f32x4 acc[4][4];
#pragma unroll
  for (int m = 0; m < 4; ++m)
#pragma unroll
    for (int n = 0; n < 4; ++n) acc[m][n] = (f32x4){0.f, 0.f, 0.f, 0.f};

  const int mstart = (kblk * 256) / 64;  // first m-tile any wave needs

  for (int t = mstart; t < N / 64; ++t) {
    __syncthreads();
    {
      // t tile: one 16-B write per unit into the linear [m][d] image
#pragma unroll
      for (int pass = 0; pass < 2; ++pass) {
        const int u = pass * SGU_BLOCK + (int)threadIdx.x;
        const int m = u >> 3;
        const int dd = (u & 7) * 8;
        bf16x8 v = *(const bf16x8*)(t_in + bND + (long long)(t * 64 + m) * D +
                                    d0 + dd);
        *(bf16x8*)(tt_lds + m * 128 + ((dd * 2) ^ (uk128(m) * 32))) = v;
      }
    }
    {
      // W tile (all 256 k for the 4 waves), triu mask at the write:
      // one masked 16-B write per unit into the linear [m][k] image
#pragma unroll
      for (int it = 0; it < 8; ++it) {
        const int u = it * SGU_BLOCK + (int)threadIdx.x;
        const int mm = u >> 5;           // 0..63
        const int k8 = (u & 31) * 8;     // 0..255 step 8
        const int m = t * 64 + mm;
        const int kglob = kblk * 256 + k8;
        bf16x8 wv = *(const bf16x8*)(w + (long long)m * N + kglob);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          if (m < kglob + j) ((short*)&wv)[j] = 0;
        *(bf16x8*)(wt_lds + mm * 512 + ((k8 * 2) ^ (uk512(mm) * 32))) = wv;
      }
    }
    __syncthreads();

    if (t * 64 >= k0) {
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        const int r1 = ks * 32 + l4 * 8 + (l15 >> 2);
        const int r2 = r1 + 4;
        // B-fragments (t image, [d][m-contig]) hoisted: reused by the
        // 4 A-fragments
        bf16x8 bfr[4];
#pragma unroll
        for (int n = 0; n < 4; ++n) {
          const int cb = (n * 16 + (l15 & 3) * 4) * 2;
          bfr[n] = sgu_frag_tr(tt_lds, 128, r1, r2,
                               cb ^ (uk128(r1) * 32), cb ^ (uk128(r2) * 32));
        }
#pragma unroll
        for (int m = 0; m < 4; ++m) {
          const int cb = (wid * 64 + m * 16 + (l15 & 3) * 4) * 2;
          bf16x8 af = sgu_frag_tr(wt_lds, 512, r1, r2,
                                  cb ^ (uk512(r1) * 32), cb ^ (uk512(r2) * 32));
#pragma unroll
          for (int n = 0; n < 4; ++n)
            acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                af, bfr[n], acc[m][n], 0, 0, 0);
        }
      }
    }
  }

#pragma unroll
  for (int m = 0; m < 4; ++m)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = k0 + m * 16 + l4 * 4 + r;
#pragma unroll
      for (int n = 0; n < 4; ++n) {
        const int d = d0 + n * 16 + l15;
        dg[bND + (long long)row * D + d] = f2bf(((float*)&acc[m][n])[r]);
      }
    }
}

// ---------------------------------------------------------------------------
// dW[m,k] = sum_{b,d} t[b,m,d] g_ln[b,k,d]  (k <= m)
// grid: (tri_tile, d_chunk, batch); fragment-shaped global loads (both
// operands naturally k-contiguous); fp32 atomicAdd combine over chunks.
// ---------------------------------------------------------------------------

#define DW_DCHUNK 768

__global__ __launch_bounds__(WAVE) void sgu_dw_kernel(
    const short* __restrict__ t_in,  // (B, N, D)
    const short* __restrict__ g_ln,  // (B, N, D)
    float* __restrict__ dw,          // (n, n) fp32, zero-init
    const int* __restrict__ tri_m,   // tri tile list: m-tile index
    const int* __restrict__ tri_k,   // tri tile list: k-tile index
    int B, int N, int D) {
  const int mt = tri_m[blockIdx.x];
  const int kt = tri_k[blockIdx.x];
  const int dstart = blockIdx.y * DW_DCHUNK;
  const int batch = blockIdx.z;
  const int dend = min(dstart + DW_DCHUNK, D);

  const int lane = threadIdx.x;
  const int l15 = lane & 15;
  const int l4 = lane >> 4;
  const long long bND = (long long)batch * N * D;

  f32x4 acc[4][4];
#pragma unroll
  for (int m = 0; m < 4; ++m)
#pragma unroll
    for (int n = 0; n < 4; ++n) acc[m][n] = (f32x4){0.f, 0.f, 0.f, 0.f};

  for (int dd = dstart; dd < dend; dd += 32) {
    const int kk0 = dd + 8 * l4;
#pragma unroll
    for (int m = 0; m < 4; ++m) {
      const int mrow = mt * 64 + m * 16 + l15;
      bf16x8 af = *(const bf16x8*)(t_in + bND + (long long)mrow * D + kk0);
#pragma unroll
      for (int n = 0; n < 4; ++n) {
        const int krow = kt * 64 + n * 16 + l15;
        bf16x8 bf = *(const bf16x8*)(g_ln + bND + (long long)krow * D + kk0);
        acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf, acc[m][n],
                                                            0, 0, 0);
      }
    }
  }

  const bool diag = (mt == kt);
#pragma unroll
  for (int m = 0; m < 4; ++m)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int mrow = mt * 64 + m * 16 + l4 * 4 + r;
#pragma unroll
      for (int n = 0; n < 4; ++n) {
        const int krow = kt * 64 + n * 16 + l15;
        if (!diag || krow <= mrow)
          atomicAdd(dw + (long long)mrow * N + krow, ((float*)&acc[m][n])[r]);
      }
    }
}

// ---------------------------------------------------------------------------
// launchers
// ---------------------------------------------------------------------------

extern "C" {

void sgu_fwd_launch(const void* xa, const void* g_ln, const void* w,
                    const float* bias, void* out, void* gate_out, int B,
                    int N, int D, hipStream_t stream) {
  dim3 grid(N / 256, D / 64, B), block(SGU_BLOCK);
  sgu_fwd_kernel<<<grid, block, 8192, stream>>>(
      (const short*)xa, (const short*)g_ln, (const short*)w, bias,
      (short*)out, (short*)gate_out, B, N, D);
}

void sgu_dgate_launch(const void* t_in, const void* w, void* dg, int B, int N,
                      int D, hipStream_t stream) {
  dim3 grid(N / 256, D / 64, B), block(SGU_BLOCK);
  size_t lds = 8192 + 32768;  // linear [m][d] t image + [m][k] W image
  sgu_dgate_kernel<<<grid, block, lds, stream>>>(
      (const short*)t_in, (const short*)w, (short*)dg, B, N, D);
}

void sgu_dw_launch(const void* t_in, const void* g_ln, float* dw,
                   const int* tri_m, const int* tri_k, int ntri, int B, int N,
                   int D, hipStream_t stream) {
  int dchunks = (D + DW_DCHUNK - 1) / DW_DCHUNK;
  dim3 grid(ntri, dchunks, B), block(WAVE);
  sgu_dw_kernel<<<grid, block, 0, stream>>>(
      (const short*)t_in, (const short*)g_ln, dw, tri_m, tri_k, B, N, D);
}

}  // extern "C"
