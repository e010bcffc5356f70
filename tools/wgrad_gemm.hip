// Weight-gradient GEMM probe: dW[M][N] = sum_k dY[k][m] * X[k][n].
//
// Round-2 perf item (TODO.md): the step's wgrad GEMMs (K = B*N = 65536,
// M/N = 1536..6144 for ProGen-1.2B) run at 0.75-1.2 PF/s through
// hipBLASLt vs 1.4-2.0 on forward shapes. This standalone probe is the
// vehicle for beating that: a split-K MFMA kernel with the tile loop,
// staging and reduction structure in place, written with idioms already
// validated in the production kernels (XOR-swizzled LDS images,
// transpose-in-staging writes as in attention_bwd's kt_lds, fp32
// atomics as in sgu_dw), so round 2 iterates on the measured hot spots
// (glds staging, 256^2 8-phase schedule per the guide) instead of
// starting from scratch.
//
// Both operands are K-MAJOR in memory (dY: (K, M) row-major, X: (K, N)
// row-major — exactly what autograd hands the wgrad: activations and
// output-grads with the token dim leading). The MFMA fragments need
// k-contiguous runs at fixed m/n, so staging transposes: global rows
// (contiguous m/n) scatter into [m][k] / [n][k] LDS images.
//
// Split-K: blockIdx.z = K-chunk; partial tiles atomically added into an
// fp32 accumulator (zeroed by the host); a final cast kernel emits bf16.
// With S=1 the tile is written directly (no atomics).
//
// Build/run (GPU box):
//   hipcc --offload-arch=gfx950 -O3 -std=c++17 [-DVARIANT=1] \
//       tools/wgrad_gemm.hip -o /tmp/wg && /tmp/wg 4608 1536 65536 8 50
//   (args: M N K splits iters; prints TF/s + spot-check vs CPU dots)
//
// VARIANT 0: synchronous staging (stage -> barrier -> mfma -> barrier)
// VARIANT 1: double-buffered staging with early loads (the validated
//            attention_fwd T14 split: global loads for chunk t+1 issue
//            before chunk t's MFMAs, LDS writes go to the alternate
//            buffer after them — one barrier per chunk)
//
// Round-2 upgrade path (derivation done, needs on-GPU iteration):
//   glds (`global_load_lds` 16B) CANNOT build the transposed [m][k]
//   image — its LDS destination is wave-uniform base + lane*16 and the
//   transposed slot order would need an 8-element k-strided global
//   GATHER per lane (one address per lane loads 16 CONTIGUOUS bytes).
//   The guide's recipe for k-major operands is therefore: stage the
//   slab LINEARLY with glds as a blocked [k/8][m/16][8][16] image
//   (each 8x16 sub-tile contiguous, 256 B) and read fragments with
//   `ds_read_b64_tr_b16` (lane l, elem j reads element
//   (l&15) + j*16 + (l>>4)*64 past the sub-tile base: per 16-lane
//   group a transposed 4x16 strip; two tr reads build one 8-deep k
//   fragment). That replaces this file's scatter-write staging (the
//   ds_write pass and its VGPRs) with async copies — the guide measured
//   +67% from 16B glds alone on the 128^2 GEMM ladder. Verify with the
//   spot-check harness below before trusting any tr mapping.

#ifndef VARIANT
#define VARIANT 0
#endif

#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>
#include <cmath>
#include <vector>
#include <cstring>
#include "../progen_amd/ops/hip/common.h"

#define BM 128
#define BN 128
#define BK 64
#define WAVES 4           // 2x2 -> each wave owns a 64x64 quadrant
#define BLOCK (WAVES * WAVE)

__device__ __forceinline__ int swz(int row, int byte_in_row) {
  return (byte_in_row ^ ((row & 7) << 4));
}

// staging of a (BK x BM) k-major global slab into an [m][k] bf16 LDS
// image (transpose-in-staging: contiguous global reads along m,
// scattered column writes — the attention_bwd kt_lds pattern), split
// into load-to-registers / write-to-LDS halves so VARIANT 1 can issue
// the loads a chunk early (T14)
#define STAGE_PASSES ((BM * BK) / (BLOCK * 8))

__device__ __forceinline__ void stage_load(const short* __restrict__ src,
                                           long long ld, long long k0,
                                           int m0, int mspan,
                                           bf16x8* regs) {
#pragma unroll
  for (int p = 0; p < STAGE_PASSES; ++p) {
    const int idx = (p * BLOCK + (int)threadIdx.x) * 8;
    const int k = idx / mspan;
    const int m = idx % mspan;
    regs[p] = *(const bf16x8*)(src + (k0 + k) * ld + m0 + m);
  }
}

__device__ __forceinline__ void stage_write(const bf16x8* regs, int mspan,
                                            char* dst) {
#pragma unroll
  for (int p = 0; p < STAGE_PASSES; ++p) {
    const int idx = (p * BLOCK + (int)threadIdx.x) * 8;
    const int k = idx / mspan;
    const int m = idx % mspan;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int mm = m + j;
      *(short*)(dst + mm * (BK * 2) + swz(mm, k * 2)) = ((short*)&regs[p])[j];
    }
  }
}

__device__ __forceinline__ void stage_T(const short* __restrict__ src,
                                        long long ld, long long k0, int m0,
                                        int mspan, char* dst) {
  bf16x8 regs[STAGE_PASSES];
  stage_load(src, ld, k0, m0, mspan, regs);
  stage_write(regs, mspan, dst);
}

__global__ __launch_bounds__(BLOCK) void wgrad_kernel(
    const short* __restrict__ dy,  // (K, M) bf16
    const short* __restrict__ x,   // (K, N) bf16
    float* __restrict__ dw_acc,    // (M, N) fp32 (zeroed when splits>1)
    int M, int N, long long K, int splits) {
  const int ntiles_n = N / BN;
  // XCD-aware tile remap (8 XCDs, bijective form)
  const int nwg = (M / BM) * ntiles_n;
  const int orig = blockIdx.x;
  const int xcd = orig % 8, q = nwg / 8, r = nwg % 8;
  const int wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q)
               + orig / 8;
  const int tm = wg / ntiles_n, tn = wg % ntiles_n;

  const long long kchunk = (K / splits);
  const long long k_lo = blockIdx.z * kchunk;
  const long long k_hi = (blockIdx.z == splits - 1) ? K : k_lo + kchunk;

  const int lane = threadIdx.x % WAVE;
  const int wid = threadIdx.x / WAVE;
  const int l15 = lane & 15;
  const int l4 = lane >> 4;
  const int wm = (wid / 2) * 64;  // wave quadrant origin in the tile
  const int wn = (wid % 2) * 64;

  extern __shared__ __attribute__((aligned(16))) char smem[];
#if VARIANT == 1
  // double-buffered: two [BM][BK]+[BN][BK] sets, 64 KiB total
  const long long nsteps = (k_hi - k_lo) / BK;
#endif

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = (f32x4){0.f, 0.f, 0.f, 0.f};

#if VARIANT == 1
  bf16x8 aregs[STAGE_PASSES], bregs[STAGE_PASSES];
  {
    // prologue: stage chunk 0 into buffer 0
    char* a0 = smem;
    char* b0 = smem + BM * BK * 2;
    stage_load(dy, M, k_lo, tm * BM, BM, aregs);
    stage_write(aregs, BM, a0);
    stage_load(x, N, k_lo, tn * BN, BN, bregs);
    stage_write(bregs, BN, b0);
  }
  __syncthreads();
  for (long long t = 0; t < nsteps; ++t) {
    const long long k0 = k_lo + t * BK;
    char* base = smem + (t & 1) * (BM + BN) * BK * 2;
    char* a_lds = base;
    char* b_lds = base + BM * BK * 2;
    if (t + 1 < nsteps) {  // T14: issue next chunk's loads before MFMAs
      stage_load(dy, M, k0 + BK, tm * BM, BM, aregs);
      stage_load(x, N, k0 + BK, tn * BN, BN, bregs);
    }
#else
  char* a_lds = smem;
  char* b_lds = smem + BM * BK * 2;
  for (long long k0 = k_lo; k0 < k_hi; k0 += BK) {
    stage_T(dy, M, k0, tm * BM, BM, a_lds);
    stage_T(x, N, k0, tn * BN, BN, b_lds);
    __syncthreads();
#endif

    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      // A-frags for this wave's 4 row blocks, B-frags for 4 col blocks
      bf16x8 af[4], bf[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int m = wm + i * 16 + l15;
        af[i] = *(const bf16x8*)(a_lds + m * (BK * 2) +
                                 swz(m, (ks * 32 + 8 * l4) * 2));
      }
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int n = wn + j * 16 + l15;
        bf[j] = *(const bf16x8*)(b_lds + n * (BK * 2) +
                                 swz(n, (ks * 32 + 8 * l4) * 2));
      }
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[i], bf[j], acc[i][j], 0, 0, 0);
    }
    __builtin_amdgcn_s_setprio(0);
#if VARIANT == 1
    if (t + 1 < nsteps) {  // write next chunk into the alternate buffer
      char* nbase = smem + ((t + 1) & 1) * (BM + BN) * BK * 2;
      stage_write(aregs, BM, nbase);
      stage_write(bregs, BN, nbase + BM * BK * 2);
    }
    __syncthreads();
  }
#else
    __syncthreads();
  }
#endif

  // epilogue: C layout — col = lane&15, row = (lane>>4)*4 + reg
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int r4 = 0; r4 < 4; ++r4) {
      const int m = tm * BM + wm + i * 16 + l4 * 4 + r4;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int n = tn * BN + wn + j * 16 + l15;
        const float v = ((float*)&acc[i][j])[r4];
        if (splits > 1) atomicAdd(dw_acc + (long long)m * N + n, v);
        else dw_acc[(long long)m * N + n] = v;
      }
    }
}

__global__ void cast_bf16_kernel(const float* __restrict__ src,
                                 short* __restrict__ dst, long long n) {
  long long i = blockIdx.x * 256LL + threadIdx.x;
  if (i < n) dst[i] = f2bf(src[i]);
}

int main(int argc, char** argv) {
  int M = argc > 1 ? atoi(argv[1]) : 4608;
  int N = argc > 2 ? atoi(argv[2]) : 1536;
  long long K = argc > 3 ? atoll(argv[3]) : 65536;
  const char* slist = argc > 4 ? argv[4] : "8";  // comma-separated splits
  int iters = argc > 5 ? atoi(argv[5]) : 50;
  if (M % BM || N % BN || K % BK) {
    printf("shape must divide: M%%%d N%%%d K%%%d\n", BM, BN, BK);
    return 1;
  }

  std::vector<short> ha((size_t)K * M), hb((size_t)K * N);
  // fast xorshift fill (host rand() is too slow for GB-scale buffers)
  auto fill = [](std::vector<short>& v, unsigned seed) {
    unsigned s = seed;
    for (auto& e : v) {
      s ^= s << 13; s ^= s >> 17; s ^= s << 5;
      float f = ((float)(s & 0xffffff) / 8388608.0f - 1.0f);  // [-1,1)
      union { float f; unsigned u; } c; c.f = f;
      e = (short)(c.u >> 16);
    }
  };
  fill(ha, 3u);
  fill(hb, 77u);

  short *da, *db, *dout;
  float* dacc;
  hipMalloc(&da, (size_t)K * M * 2);
  hipMalloc(&db, (size_t)K * N * 2);
  hipMalloc(&dacc, (size_t)M * N * 4);
  hipMalloc(&dout, (size_t)M * N * 2);
  hipMemcpy(da, ha.data(), (size_t)K * M * 2, hipMemcpyHostToDevice);
  hipMemcpy(db, hb.data(), (size_t)K * N * 2, hipMemcpyHostToDevice);

#if VARIANT == 1
  size_t lds = 2 * (size_t)(BM + BN) * BK * 2;
#else
  size_t lds = (size_t)(BM + BN) * BK * 2;
#endif

  int rc = 0;
  char sbuf[256];
  snprintf(sbuf, sizeof sbuf, "%s", slist);
  for (char* tok = strtok(sbuf, ","); tok; tok = strtok(nullptr, ",")) {
    int S = atoi(tok);
    if (K % (long long)(BK * S)) { printf("skip S=%d (K)\n", S); continue; }
    dim3 grid((M / BM) * (N / BN), 1, S), block(BLOCK);

    auto run = [&]() {
      if (S > 1) hipMemsetAsync(dacc, 0, (size_t)M * N * 4);
      wgrad_kernel<<<grid, block, lds>>>(da, db, dacc, M, N, K, S);
      long long n = (long long)M * N;
      cast_bf16_kernel<<<(int)((n + 255) / 256), 256>>>(dacc, dout, n);
    };

    for (int i = 0; i < 5; ++i) run();
    hipDeviceSynchronize();
    hipError_t err = hipGetLastError();
    if (err != hipSuccess) { printf("HIP ERR %s\n", hipGetErrorString(err)); return 1; }

    hipEvent_t e0, e1;
    hipEventCreate(&e0); hipEventCreate(&e1);
    hipEventRecord(e0);
    for (int i = 0; i < iters; ++i) run();
    hipEventRecord(e1);
    hipEventSynchronize(e1);
    float ms;
    hipEventElapsedTime(&ms, e0, e1);
    double us = ms * 1000.0 / iters;
    double tf = 2.0 * M * N * (double)K / (us * 1e-6) / 1e12;
    printf("VARIANT %d wgrad %dx%dx%lld S=%d: %.1f us  %.1f TF/s\n",
           VARIANT, M, N, K, S, us, tf);

    // spot-check ~64 random outputs against fp32 CPU dots over K
    std::vector<float> got((size_t)M * N);
    hipMemcpy(got.data(), dacc, (size_t)M * N * 4, hipMemcpyDeviceToHost);
    auto b2f = [](short s) {
      union { unsigned u; float f; } c; c.u = ((unsigned)(unsigned short)s) << 16;
      return c.f;
    };
    double maxrel = 0;
    for (int t = 0; t < 64; ++t) {
      int m = rand() % M, n = rand() % N;
      double ref = 0;
      for (long long k = 0; k < K; ++k)
        ref += (double)b2f(ha[(size_t)k * M + m]) * (double)b2f(hb[(size_t)k * N + n]);
      double g = got[(size_t)m * N + n];
      double rel = fabs(g - ref) / (fabs(ref) + 1e-3);
      if (rel > maxrel) maxrel = rel;
    }
    printf("spot-check max rel err (64 samples): %.3e %s\n", maxrel,
           maxrel < 2e-2 ? "OK" : "FAIL");
    if (maxrel >= 2e-2) rc = 1;
  }
  hipFree(da); hipFree(db); hipFree(dacc); hipFree(dout);
  return rc;
}
