// Weight-gradient GEMM, generation 2: glds + ds_read_b64_tr_b16.
//
// dW[M][N] = sum_k dY[k][m] * X[k][n]; both operands K-MAJOR in memory
// (dY: (K, M) row-major, X: (K, N) row-major — what autograd hands the
// wgrad). Generation 1 (tools/wgrad_gemm.hip, scatter-write register
// staging, 128^2 tile) measured 115-148 TF/s vs hipBLASLt's 0.94-1.08
// PF/s on the four ProGen-1.2B shapes — instruction-bound on its 16
// b16 LDS scatter writes per 16-B load. This file is the guide's
// recipe for k-major operands instead:
//
//   - 256x256 tile, BK=64, 8 waves (2M x 4N, per-wave 128x64 output);
//   - operands staged LINEARLY with global_load_lds (16 B/lane, no
//     ds_write pass, no staging VGPRs) into [k][512 B] row images,
//     double-buffered (128 KiB LDS);
//   - MFMA fragments read with ds_read_b64_tr_b16 (hardware 4x16
//     transpose): lane l of a 16-lane group receives column (l&15) of
//     the 4x16 row-major block assembled from the group's 16 8-B
//     chunks (row j = lanes 4j..4j+3, addresses per-lane => row
//     placement is free);
//   - bank conflicts: a 32-lane half reads 8 k-rows x 8 dwords; rows
//     are 512 B apart (== bank 0 mod 64 dwords), so within-row XOR
//     swizzle col_bytes ^= u(k)*32 with u(k) = (k&3)|((k&8)>>1) places
//     the half-wave's 8 rows ({kb..kb+3, kb+8..kb+11}) in 8 distinct
//     8-dword windows -> zero-conflict; u depends only on k&0xb so the
//     XOR granularity (32 B) preserves the 16-B glds chunks and the
//     8-B tr chunks;
//   - split-K over blockIdx.z, fp32 atomics into dw_acc, bf16 cast
//     kernel (same ending as gen 1).
//
// VARIANT 0: 2 LDS buffers, glds for chunk t+1 issued before compute
//            of chunk t, vmcnt(0) + plain __syncthreads() per K-step
//            (the guide's "glds, 2 buffers, BK=64" configuration).
// VARIANT 1: 2 buffers, counted vmcnt across raw s_barriers keeps one
//            tile's glds in flight (the 8-phase template's sync
//            scheme, 2-deep).
// VARIANT 2: ABLATION — stage tile 0 once, loop MFMAs+barriers over it
//            (wrong results; isolates the MFMA+barrier skeleton time).
// VARIANT 3: ABLATION — staging+sync only, MFMAs skipped with the
//            fragments kept alive via asm (wrong results; isolates the
//            staging/HBM side).
// VARIANT 8: the 8-phase template's fine interleave (guide: 'the
//            per-phase interleave is the lever', -7..27% without it).
//            Each K-step = 4 phases of [16 tr reads | 2 glds of one
//            half of tile t+1 | raw barrier | lgkmcnt(0) | 16 MFMA];
//            staging order A0,B0,A1,B1 matches the next step's read
//            order so counted vmcnt(4) at phases 0 and 2 suffices.
//            2 full buffers (ping-pong), stage depth 1 tile.
// VARIANT 7: VARIANT 6 + ONE barrier per K-step (the counted
//            vmcnt(4) moves BEFORE the barrier, so the same barrier
//            orders both "reads of t done" and "tile t+1 landed in
//            every wave"; B(t+2)'s glds issue after it) + split
//            read/MFMA phases per ks: the A mb4-7 tr reads issue
//            before the mb0-3 MFMA cluster and land under it
//            (lgkmcnt(8) leaves exactly them outstanding).
// VARIANT 6: VARIANT 5 + A-operand 3-slot LDS ring (96 KiB A + 64 KiB
//            B = 160 KiB, the full CU LDS): slot (t+2)%3 is free
//            DURING compute of step t (its readers finished at step
//            t-1's barrier), so the A-half of the next-next tile's
//            glds issues mid-compute, hidden under the ks=0 MFMA
//            cluster, leaving only B's 4 glds + the counted wait in
//            the barrier gap.
// VARIANT 5: hand-scheduled pipeline. The round-2 PMC + .s inspection
//            found hipcc emits `s_waitcnt vmcnt(0)` before the FIRST
//            ds_read of every k-step whenever a glds is in flight (the
//            guide's §6 trap), draining the prefetched tile and making
//            VARIANT 1's counted-vmcnt scheme identical to VARIANT 0
//            (both measured 860 TF/s vs the 1463 TF/s compute-only
//            skeleton). Here the tr reads are INLINE ASM (invisible to
//            hipcc's alias analysis, so it stops injecting the drain),
//            with manual `s_waitcnt lgkmcnt(0)` + sched_barrier before
//            each MFMA cluster, raw s_barriers, and a counted
//            vmcnt(8) that keeps one tile's glds in flight across the
//            barrier pair.
//
// Build/run (GPU box):
//   hipcc --offload-arch=gfx950 -O3 -std=c++17 [-DVARIANT=1] \
//       tools/wgrad_gemm2.hip -o /tmp/wg2 && /tmp/wg2 4608 1536 65536 8,16 30
// Runs a tr-mapping self-test first (validates the lane->element
// theory on device before any timing).

#ifndef VARIANT
#define VARIANT 0
#endif

#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>
#include <cmath>
#include <cstring>
#include <vector>
#include "../progen_amd/ops/hip/common.h"

#define BM 256
#define BN 256
#define BK 64
#define NWAVES 8
#define BLOCK (NWAVES * WAVE)

typedef __attribute__((__vector_size__(4 * sizeof(__bf16)))) __bf16 bf16x4t;
#define AS1 __attribute__((address_space(1)))
#define AS3 __attribute__((address_space(3)))

__device__ __forceinline__ int uk(int k) { return (k & 3) | ((k & 8) >> 1); }

// stage one BK x 256-col k-major slab into a [k][512 B] LDS image with
// the 32-B XOR swizzle, via glds: 4 wave-instructions per wave (32
// total, 2 k-rows each). src points at (k-row 0, tile col 0); ldb =
// global row stride in bytes.
__device__ __forceinline__ void stage_glds(const char* __restrict__ src,
                                           long long ldb, char* img) {
  const int lane = (int)threadIdx.x & 63;
  const int wid = (int)threadIdx.x >> 6;
#pragma unroll
  for (int p = 0; p < 4; ++p) {
    const int inst = p * NWAVES + wid;
    const int krow = inst * 2 + (lane >> 5);
    const int colb = ((lane & 31) * 16) ^ (uk(krow) * 32);
    __builtin_amdgcn_global_load_lds(
        (const AS1 unsigned int*)(src + (long long)krow * ldb + colb),
        (AS3 unsigned int*)(img + inst * 1024), 16, 0, 0);
  }
}

// MFMA fragment (8 k x 16 cols) from a staged image via two tr reads.
// kb = k base (0 or 32), m0 = column base within the 256-col image.
__device__ __forceinline__ bf16x8 frag_tr(const char* img, int kb, int m0,
                                          int l15, int l4) {
  const int r1 = kb + l4 * 8 + (l15 >> 2);
  const int r2 = r1 + 4;
  const int colb = (m0 + (l15 & 3) * 4) * 2;
  auto p1 = (AS3 bf16x4t*)(img + r1 * 512 + (colb ^ (uk(r1) * 32)));
  auto p2 = (AS3 bf16x4t*)(img + r2 * 512 + (colb ^ (uk(r2) * 32)));
  bf16x4t a = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p1);
  bf16x4t b = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p2);
  bf16x8 o;
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    ((__bf16*)&o)[j] = a[j];
    ((__bf16*)&o)[j + 4] = b[j];
  }
  return o;
}

#if VARIANT >= 5
// one 8-bf16 MFMA A/B fragment from two inline-asm tr reads at integer
// LDS byte offsets (compiler cannot see these as LDS reads)
union FragU {
  bf16x8 v;
  struct { unsigned long long lo, hi; } u;
};

__device__ __forceinline__ bf16x8 frag_tr_asm(unsigned off1, unsigned off2) {
  FragU f;
  // early-clobber (=&v) is required: instruction 1's destination must
  // not be allocated over instruction 2's address operand (%3) — plain
  // "=v" produced exactly that aliasing and garbage fragments
  asm volatile("ds_read_b64_tr_b16 %0, %2\n\tds_read_b64_tr_b16 %1, %3"
               : "=&v"(f.u.lo), "=&v"(f.u.hi)
               : "v"(off1), "v"(off2));
  return f.v;
}

__global__ __launch_bounds__(BLOCK) void wgrad2_v5_kernel(
    const short* __restrict__ dy, const short* __restrict__ x,
    float* __restrict__ dw_acc, int M, int N, long long K, int splits) {
  const int ntiles_n = N / BN;
  const int nwg = (M / BM) * ntiles_n;
  const int orig = blockIdx.x;
  const int xcd = orig % 8, q = nwg / 8, r = nwg % 8;
  const int wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q)
               + orig / 8;
  const int tm = wg / ntiles_n, tn = wg % ntiles_n;

  const long long kchunk = K / splits;
  const long long k_lo = blockIdx.z * kchunk;
  const long long nsteps = kchunk / BK;

  const int lane = (int)threadIdx.x & 63;
  const int l15 = lane & 15;
  const int l4 = lane >> 4;
  const int wid = (int)threadIdx.x >> 6;
  const int wr = wid >> 2;
  const int wc = wid & 3;

  const long long lda = 2LL * M;
  const long long ldb2 = 2LL * N;
  const char* asrc = (const char*)dy + k_lo * lda + (long long)tm * 512;
  const char* bsrc = (const char*)x + k_lo * ldb2 + (long long)tn * 512;

  extern __shared__ __attribute__((aligned(16))) char smem[];

  // per-lane row constants: rows r(ks,j) = ks*32 + l4*8 + (l15>>2) + 4j
  // (j=0,1). The column offset must be XORed as a WHOLE with u(row)*32
  // (bits 5..7): wc*128 carries bit 7, so it CANNOT be hoisted out of
  // the XOR additively (the first V5 did, and B-frags with wc in {1,3}
  // and u >= 4 read garbage — spot-check FAIL). arow/brow hold only the
  // XOR-neutral parts (row base, bits 0..4 and >= 8-safe A term).
  unsigned arowb[2][2], browb[2][2], au32[2][2];
  const unsigned lp = (unsigned)((l15 & 3) * 8);
#pragma unroll
  for (int ks = 0; ks < 2; ++ks)
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      const int rr = ks * 32 + l4 * 8 + (l15 >> 2) + 4 * j;
      au32[ks][j] = (unsigned)(uk(rr) * 32);
      arowb[ks][j] = (unsigned)(rr * 512);
      browb[ks][j] = (unsigned)(32768 + rr * 512);
    }
  const unsigned acol = (unsigned)(wr * 256) + lp;  // + mb*32, then ^u32
  const unsigned bcol = (unsigned)(wc * 128) + lp;  // + nb*32, then ^u32

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = (f32x4){0.f, 0.f, 0.f, 0.f};

  {
    stage_glds(asrc, lda, smem);
    stage_glds(bsrc, ldb2, smem + 32768);
    if (nsteps > 1) {
      stage_glds(asrc + BK * lda, lda, smem + 65536);
      stage_glds(bsrc + BK * ldb2, ldb2, smem + 65536 + 32768);
    }
  }
  asm volatile("s_waitcnt vmcnt(8)" ::: "memory");  // tile 0 landed (mine)
  __builtin_amdgcn_s_barrier();                     // ...and everyone's

  for (long long t = 0; t < nsteps; ++t) {
    const unsigned tb = (unsigned)((t & 1) * 65536);
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      bf16x8 bfr[4], afr[8];
#pragma unroll
      for (int nb = 0; nb < 4; ++nb)
        bfr[nb] = frag_tr_asm(
            tb + browb[ks][0] + ((bcol + 32 * nb) ^ au32[ks][0]),
            tb + browb[ks][1] + ((bcol + 32 * nb) ^ au32[ks][1]));
#pragma unroll
      for (int mb = 0; mb < 8; ++mb)
        afr[mb] = frag_tr_asm(
            tb + arowb[ks][0] + ((acol + 32 * mb) ^ au32[ks][0]),
            tb + arowb[ks][1] + ((acol + 32 * mb) ^ au32[ks][1]));
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_sched_barrier(0);  // §5.4 rule 18: pin MFMAs after
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int mb = 0; mb < 8; ++mb)
#pragma unroll
        for (int nb = 0; nb < 4; ++nb)
          acc[mb][nb] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afr[mb], bfr[nb], acc[mb][nb], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
    __builtin_amdgcn_s_barrier();  // raw: buffer t&1 reads done everywhere
    if (t + 2 < nsteps) {
      const long long koff = (t + 2) * BK;
      char* nxt = smem + (t & 1) * 65536;
      stage_glds(asrc + koff * lda, lda, nxt);
      stage_glds(bsrc + koff * ldb2, ldb2, nxt + 32768);
      asm volatile("s_waitcnt vmcnt(8)" ::: "memory");  // t+1 landed (mine)
    } else if (t + 1 < nsteps) {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    if (t + 1 < nsteps)
      __builtin_amdgcn_s_barrier();  // t+1 landed in every wave's rows
  }

  const long long mbase = (long long)tm * BM + wr * 128;
  const long long nbase = (long long)tn * BN + wc * 64;
#pragma unroll
  for (int mb = 0; mb < 8; ++mb)
#pragma unroll
    for (int r4 = 0; r4 < 4; ++r4) {
      const long long m = mbase + mb * 16 + l4 * 4 + r4;
#pragma unroll
      for (int nb = 0; nb < 4; ++nb) {
        const long long n = nbase + nb * 16 + l15;
        const float v = ((float*)&acc[mb][nb])[r4];
        if (splits > 1) atomicAdd(dw_acc + m * N + n, v);
        else dw_acc[m * N + n] = v;
      }
    }
}
#endif  // VARIANT == 5




#if VARIANT == 8
// stage HALF a slab (32 k-rows): 2 wave-instructions per wave
__device__ __forceinline__ void stage_glds_half(const char* __restrict__ src,
                                                long long ldb, char* img,
                                                int khalf) {
  const int lane = (int)threadIdx.x & 63;
  const int wid = (int)threadIdx.x >> 6;
#pragma unroll
  for (int p = 0; p < 2; ++p) {
    const int inst = khalf * 16 + p * NWAVES + wid;
    const int krow = inst * 2 + (lane >> 5);
    const int colb = ((lane & 31) * 16) ^ (uk(krow) * 32);
    __builtin_amdgcn_global_load_lds(
        (const AS1 unsigned int*)(src + (long long)krow * ldb + colb),
        (AS3 unsigned int*)(img + inst * 1024), 16, 0, 0);
  }
}

__global__ __launch_bounds__(BLOCK) void wgrad2_v8_kernel(
    const short* __restrict__ dy, const short* __restrict__ x,
    float* __restrict__ dw_acc, int M, int N, long long K, int splits) {
  const int ntiles_n = N / BN;
  const int nwg = (M / BM) * ntiles_n;
  const int orig = blockIdx.x;
  const int xcd = orig % 8, q = nwg / 8, r = nwg % 8;
  const int wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q)
               + orig / 8;
  const int tm = wg / ntiles_n, tn = wg % ntiles_n;

  const long long kchunk = K / splits;
  const long long k_lo = blockIdx.z * kchunk;
  const long long nsteps = kchunk / BK;

  const int lane = (int)threadIdx.x & 63;
  const int l15 = lane & 15;
  const int l4 = lane >> 4;
  const int wid = (int)threadIdx.x >> 6;
  const int wr = wid >> 2;
  const int wc = wid & 3;

  const long long lda = 2LL * M;
  const long long ldb2 = 2LL * N;
  const char* asrc = (const char*)dy + k_lo * lda + (long long)tm * 512;
  const char* bsrc = (const char*)x + k_lo * ldb2 + (long long)tn * 512;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  // [buf][A 32K | B 32K]

  unsigned arowb[2][2], brow_r[2][2], au32[2][2];
  const unsigned lp = (unsigned)((l15 & 3) * 8);
#pragma unroll
  for (int ks = 0; ks < 2; ++ks)
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      const int rr = ks * 32 + l4 * 8 + (l15 >> 2) + 4 * j;
      au32[ks][j] = (unsigned)(uk(rr) * 32);
      arowb[ks][j] = (unsigned)(rr * 512);
      brow_r[ks][j] = (unsigned)(32768 + rr * 512);
    }
  const unsigned acol = (unsigned)(wr * 256) + lp;
  const unsigned bcol = (unsigned)(wc * 128) + lp;

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = (f32x4){0.f, 0.f, 0.f, 0.f};

  {
    stage_glds(asrc, lda, smem);
    stage_glds(bsrc, ldb2, smem + 32768);
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  for (long long t = 0; t < nsteps; ++t) {
    const unsigned tb = (unsigned)((t & 1) * 65536);
    char* nxt = smem + ((t + 1) & 1) * 65536;
    const char* an = asrc + (t + 1) * BK * lda;
    const char* bn = bsrc + (t + 1) * BK * ldb2;
    bf16x8 bfr[4];
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      const int ks = p >> 1, mh = p & 1;
      bf16x8 afr[4];
      if (mh == 0) {
#pragma unroll
        for (int nb = 0; nb < 4; ++nb)
          bfr[nb] = frag_tr_asm(
              tb + brow_r[ks][0] + ((bcol + 32 * nb) ^ au32[ks][0]),
              tb + brow_r[ks][1] + ((bcol + 32 * nb) ^ au32[ks][1]));
      }
#pragma unroll
      for (int mb = 0; mb < 4; ++mb) {
        const int mbb = mh * 4 + mb;
        afr[mb] = frag_tr_asm(
            tb + arowb[ks][0] + ((acol + 32 * mbb) ^ au32[ks][0]),
            tb + arowb[ks][1] + ((acol + 32 * mbb) ^ au32[ks][1]));
      }
      // stage one half of tile t+1 (order A0,B0,A1,B1 = read order)
      if (t + 1 < nsteps) {
        if (p == 0) stage_glds_half(an, lda, nxt, 0);
        else if (p == 1) stage_glds_half(bn, ldb2, nxt + 32768, 0);
        else if (p == 2) stage_glds_half(an, lda, nxt, 1);
        else stage_glds_half(bn, ldb2, nxt + 32768, 1);
      }
      if (p == 2) {
        // phase 2 reads A1/B1 of tile t (staged at t-1's phases 2-3);
        // newer in flight: A0,B0,A1 of t+1 = 6 instructions
        if (t + 1 < nsteps)
          asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
        else
          asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      }
      __builtin_amdgcn_s_barrier();
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_sched_barrier(0);
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int mb = 0; mb < 4; ++mb)
#pragma unroll
        for (int nb = 0; nb < 4; ++nb)
          acc[mh * 4 + mb][nb] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afr[mb], bfr[nb], acc[mh * 4 + mb][nb], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
      __builtin_amdgcn_s_barrier();
    }
    // next step's phase 0 reads A0/B0 of t+1 — the FIRST 4 glds issued
    // this step; its A1/B1 (4 newer) stay in flight across the barrier
    // and are landed by the phase-2 vmcnt(6) above
    if (t + 1 < nsteps)
      asm volatile("s_waitcnt vmcnt(4)" ::: "memory");  // A0,B0 landed
    __builtin_amdgcn_s_barrier();
  }

  const long long mbase = (long long)tm * BM + wr * 128;
  const long long nbase = (long long)tn * BN + wc * 64;
#pragma unroll
  for (int mb = 0; mb < 8; ++mb)
#pragma unroll
    for (int r4 = 0; r4 < 4; ++r4) {
      const long long m = mbase + mb * 16 + l4 * 4 + r4;
#pragma unroll
      for (int nb = 0; nb < 4; ++nb) {
        const long long n = nbase + nb * 16 + l15;
        const float v = ((float*)&acc[mb][nb])[r4];
        if (splits > 1) atomicAdd(dw_acc + m * N + n, v);
        else dw_acc[m * N + n] = v;
      }
    }
}
#endif  // VARIANT == 8

#if VARIANT == 7
__global__ __launch_bounds__(BLOCK) void wgrad2_v7_kernel(
    const short* __restrict__ dy, const short* __restrict__ x,
    float* __restrict__ dw_acc, int M, int N, long long K, int splits) {
  const int ntiles_n = N / BN;
  const int nwg = (M / BM) * ntiles_n;
  const int orig = blockIdx.x;
  const int xcd = orig % 8, q = nwg / 8, r = nwg % 8;
  const int wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q)
               + orig / 8;
  const int tm = wg / ntiles_n, tn = wg % ntiles_n;

  const long long kchunk = K / splits;
  const long long k_lo = blockIdx.z * kchunk;
  const long long nsteps = kchunk / BK;

  const int lane = (int)threadIdx.x & 63;
  const int l15 = lane & 15;
  const int l4 = lane >> 4;
  const int wid = (int)threadIdx.x >> 6;
  const int wr = wid >> 2;
  const int wc = wid & 3;

  const long long lda = 2LL * M;
  const long long ldb2 = 2LL * N;
  const char* asrc = (const char*)dy + k_lo * lda + (long long)tm * 512;
  const char* bsrc = (const char*)x + k_lo * ldb2 + (long long)tn * 512;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  const unsigned ABASE[3] = {0u, 32768u, 65536u};
  const unsigned BBASE[2] = {98304u, 131072u};

  unsigned arowb[2][2], brow_r[2][2], au32[2][2];
  const unsigned lp = (unsigned)((l15 & 3) * 8);
#pragma unroll
  for (int ks = 0; ks < 2; ++ks)
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      const int rr = ks * 32 + l4 * 8 + (l15 >> 2) + 4 * j;
      au32[ks][j] = (unsigned)(uk(rr) * 32);
      arowb[ks][j] = (unsigned)(rr * 512);
      brow_r[ks][j] = (unsigned)(rr * 512);
    }
  const unsigned acol = (unsigned)(wr * 256) + lp;
  const unsigned bcol = (unsigned)(wc * 128) + lp;

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = (f32x4){0.f, 0.f, 0.f, 0.f};

  {
    stage_glds(asrc, lda, smem + ABASE[0]);
    stage_glds(bsrc, ldb2, smem + BBASE[0]);
    if (nsteps > 1) {
      stage_glds(asrc + BK * lda, lda, smem + ABASE[1]);
      stage_glds(bsrc + BK * ldb2, ldb2, smem + BBASE[1]);
    }
  }
  asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  for (long long t = 0; t < nsteps; ++t) {
    const unsigned ta = ABASE[t % 3];
    const unsigned tbb = BBASE[t & 1];
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      bf16x8 bfr[4], afr[8];
#pragma unroll
      for (int nb = 0; nb < 4; ++nb)
        bfr[nb] = frag_tr_asm(
            tbb + brow_r[ks][0] + ((bcol + 32 * nb) ^ au32[ks][0]),
            tbb + brow_r[ks][1] + ((bcol + 32 * nb) ^ au32[ks][1]));
#pragma unroll
      for (int mb = 0; mb < 4; ++mb)
        afr[mb] = frag_tr_asm(
            ta + arowb[ks][0] + ((acol + 32 * mb) ^ au32[ks][0]),
            ta + arowb[ks][1] + ((acol + 32 * mb) ^ au32[ks][1]));
      // issue the second A half now; it lands under the first cluster
#pragma unroll
      for (int mb = 4; mb < 8; ++mb)
        afr[mb] = frag_tr_asm(
            ta + arowb[ks][0] + ((acol + 32 * mb) ^ au32[ks][0]),
            ta + arowb[ks][1] + ((acol + 32 * mb) ^ au32[ks][1]));
      asm volatile("s_waitcnt lgkmcnt(8)" ::: "memory");  // halves 0-3 in
      __builtin_amdgcn_sched_barrier(0);
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int mb = 0; mb < 4; ++mb)
#pragma unroll
        for (int nb = 0; nb < 4; ++nb)
          acc[mb][nb] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afr[mb], bfr[nb], acc[mb][nb], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
      if (ks == 0 && t + 2 < nsteps)
        stage_glds(asrc + (t + 2) * BK * lda, lda,
                   smem + ABASE[(t + 2) % 3]);
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");  // halves 4-7 in
      __builtin_amdgcn_sched_barrier(0);
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int mb = 4; mb < 8; ++mb)
#pragma unroll
        for (int nb = 0; nb < 4; ++nb)
          acc[mb][nb] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afr[mb], bfr[nb], acc[mb][nb], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
    // single barrier: per-wave vmcnt(4) first (A(t+1)/B(t+1) landed;
    // A(t+2)'s 4 glds stay in flight), then the barrier orders BOTH
    // "everyone done reading t" and "t+1 landed everywhere"
    if (t + 2 < nsteps) {
      asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();
    if (t + 2 < nsteps)
      stage_glds(bsrc + (t + 2) * BK * ldb2, ldb2, smem + BBASE[t & 1]);
  }

  const long long mbase = (long long)tm * BM + wr * 128;
  const long long nbase = (long long)tn * BN + wc * 64;
#pragma unroll
  for (int mb = 0; mb < 8; ++mb)
#pragma unroll
    for (int r4 = 0; r4 < 4; ++r4) {
      const long long m = mbase + mb * 16 + l4 * 4 + r4;
#pragma unroll
      for (int nb = 0; nb < 4; ++nb) {
        const long long n = nbase + nb * 16 + l15;
        const float v = ((float*)&acc[mb][nb])[r4];
        if (splits > 1) atomicAdd(dw_acc + m * N + n, v);
        else dw_acc[m * N + n] = v;
      }
    }
}
#endif  // VARIANT == 7

#if VARIANT == 6
__global__ __launch_bounds__(BLOCK) void wgrad2_v6_kernel(
    const short* __restrict__ dy, const short* __restrict__ x,
    float* __restrict__ dw_acc, int M, int N, long long K, int splits) {
  const int ntiles_n = N / BN;
  const int nwg = (M / BM) * ntiles_n;
  const int orig = blockIdx.x;
  const int xcd = orig % 8, q = nwg / 8, r = nwg % 8;
  const int wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q)
               + orig / 8;
  const int tm = wg / ntiles_n, tn = wg % ntiles_n;

  const long long kchunk = K / splits;
  const long long k_lo = blockIdx.z * kchunk;
  const long long nsteps = kchunk / BK;

  const int lane = (int)threadIdx.x & 63;
  const int l15 = lane & 15;
  const int l4 = lane >> 4;
  const int wid = (int)threadIdx.x >> 6;
  const int wr = wid >> 2;
  const int wc = wid & 3;

  const long long lda = 2LL * M;
  const long long ldb2 = 2LL * N;
  const char* asrc = (const char*)dy + k_lo * lda + (long long)tm * 512;
  const char* bsrc = (const char*)x + k_lo * ldb2 + (long long)tn * 512;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  // A ring: slots 0/1/2 at 0,32K,64K; B pair: 96K,128K
  const unsigned ABASE[3] = {0u, 32768u, 65536u};
  const unsigned BBASE[2] = {98304u, 131072u};

  unsigned arowb[2][2], brow_r[2][2], au32[2][2];
  const unsigned lp = (unsigned)((l15 & 3) * 8);
#pragma unroll
  for (int ks = 0; ks < 2; ++ks)
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      const int rr = ks * 32 + l4 * 8 + (l15 >> 2) + 4 * j;
      au32[ks][j] = (unsigned)(uk(rr) * 32);
      arowb[ks][j] = (unsigned)(rr * 512);
      brow_r[ks][j] = (unsigned)(rr * 512);
    }
  const unsigned acol = (unsigned)(wr * 256) + lp;
  const unsigned bcol = (unsigned)(wc * 128) + lp;

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = (f32x4){0.f, 0.f, 0.f, 0.f};

  {
    stage_glds(asrc, lda, smem + ABASE[0]);
    stage_glds(bsrc, ldb2, smem + BBASE[0]);
    if (nsteps > 1) {
      stage_glds(asrc + BK * lda, lda, smem + ABASE[1]);
      stage_glds(bsrc + BK * ldb2, ldb2, smem + BBASE[1]);
    }
  }
  asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  for (long long t = 0; t < nsteps; ++t) {
    const unsigned ta = ABASE[t % 3];
    const unsigned tbb = BBASE[t & 1];
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      bf16x8 bfr[4], afr[8];
#pragma unroll
      for (int nb = 0; nb < 4; ++nb)
        bfr[nb] = frag_tr_asm(
            tbb + brow_r[ks][0] + ((bcol + 32 * nb) ^ au32[ks][0]),
            tbb + brow_r[ks][1] + ((bcol + 32 * nb) ^ au32[ks][1]));
#pragma unroll
      for (int mb = 0; mb < 8; ++mb)
        afr[mb] = frag_tr_asm(
            ta + arowb[ks][0] + ((acol + 32 * mb) ^ au32[ks][0]),
            ta + arowb[ks][1] + ((acol + 32 * mb) ^ au32[ks][1]));
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_sched_barrier(0);
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int mb = 0; mb < 8; ++mb)
#pragma unroll
        for (int nb = 0; nb < 4; ++nb)
          acc[mb][nb] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afr[mb], bfr[nb], acc[mb][nb], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
      if (ks == 0 && t + 2 < nsteps)  // A(t+2): slot free since t-1's
        stage_glds(asrc + (t + 2) * BK * lda, lda,
                   smem + ABASE[(t + 2) % 3]);  // hidden under compute
    }
    __builtin_amdgcn_s_barrier();  // raw: A[t%3]/B[t&1] reads done
    if (t + 2 < nsteps) {
      stage_glds(bsrc + (t + 2) * BK * ldb2, ldb2, smem + BBASE[t & 1]);
      asm volatile("s_waitcnt vmcnt(8)" ::: "memory");  // t+1 landed
    } else if (t + 1 < nsteps) {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    if (t + 1 < nsteps)
      __builtin_amdgcn_s_barrier();
  }

  const long long mbase = (long long)tm * BM + wr * 128;
  const long long nbase = (long long)tn * BN + wc * 64;
#pragma unroll
  for (int mb = 0; mb < 8; ++mb)
#pragma unroll
    for (int r4 = 0; r4 < 4; ++r4) {
      const long long m = mbase + mb * 16 + l4 * 4 + r4;
#pragma unroll
      for (int nb = 0; nb < 4; ++nb) {
        const long long n = nbase + nb * 16 + l15;
        const float v = ((float*)&acc[mb][nb])[r4];
        if (splits > 1) atomicAdd(dw_acc + m * N + n, v);
        else dw_acc[m * N + n] = v;
      }
    }
}
#endif  // VARIANT == 6

__global__ __launch_bounds__(BLOCK) void wgrad2_kernel(
    const short* __restrict__ dy,  // (K, M) bf16
    const short* __restrict__ x,   // (K, N) bf16
    float* __restrict__ dw_acc,    // (M, N) fp32 (zeroed when splits>1)
    int M, int N, long long K, int splits) {
  const int ntiles_n = N / BN;
  const int nwg = (M / BM) * ntiles_n;
  // XCD-aware bijective remap
  const int orig = blockIdx.x;
  const int xcd = orig % 8, q = nwg / 8, r = nwg % 8;
  const int wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q)
               + orig / 8;
  const int tm = wg / ntiles_n, tn = wg % ntiles_n;

  const long long kchunk = K / splits;
  const long long k_lo = blockIdx.z * kchunk;
  const long long nsteps = kchunk / BK;

  const int lane = (int)threadIdx.x & 63;
  const int wid = (int)threadIdx.x >> 6;
  const int l15 = lane & 15;
  const int l4 = lane >> 4;
  const int wr = wid >> 2;        // 0..1: 128-row half
  const int wc = wid & 3;         // 0..3: 64-col quarter

  const long long lda = 2LL * M;  // bytes per dY k-row
  const long long ldb2 = 2LL * N;
  const char* asrc = (const char*)dy + k_lo * lda + (long long)tm * 512;
  const char* bsrc = (const char*)x + k_lo * ldb2 + (long long)tn * 512;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  // buffer layout: [buf][A 32 KiB | B 32 KiB]

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = (f32x4){0.f, 0.f, 0.f, 0.f};

#if VARIANT == 1
  // 2 buffers, counted vmcnt across RAW barriers: tile t+2's glds are
  // issued right after the barrier that ends step t (into buffer t&1,
  // whose reads that barrier just fenced), and a counted vmcnt(8)
  // lands tile t+1 while t+2's 8 glds stay in flight across the next
  // barrier (the 8-phase template's sync scheme, 2-deep).
  {
    stage_glds(asrc, lda, smem);
    stage_glds(bsrc, ldb2, smem + 32768);
    if (nsteps > 1) {
      stage_glds(asrc + BK * lda, lda, smem + 65536);
      stage_glds(bsrc + BK * ldb2, ldb2, smem + 65536 + 32768);
    }
  }
  asm volatile("s_waitcnt vmcnt(8)" ::: "memory");  // tile 0 landed
  __builtin_amdgcn_s_barrier();
  for (long long t = 0; t < nsteps; ++t) {
    char* cur = smem + (t & 1) * 65536;
#else
  {
    stage_glds(asrc, lda, smem);
    stage_glds(bsrc, ldb2, smem + 32768);
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();
  for (long long t = 0; t < nsteps; ++t) {
#if VARIANT == 2
    char* cur = smem;  // ablation: same staged tile every step
#else
    char* cur = smem + (t & 1) * 65536;
    if (t + 1 < nsteps) {
      const long long koff = (t + 1) * BK;
      char* nxt = smem + ((t + 1) & 1) * 65536;
      stage_glds(asrc + koff * lda, lda, nxt);
      stage_glds(bsrc + koff * ldb2, ldb2, nxt + 32768);
    }
#endif
#endif

    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      bf16x8 bfr[4], afr[8];
#pragma unroll
      for (int nb = 0; nb < 4; ++nb)
        bfr[nb] = frag_tr(cur + 32768, ks * 32, wc * 64 + nb * 16, l15, l4);
#pragma unroll
      for (int mb = 0; mb < 8; ++mb)
        afr[mb] = frag_tr(cur, ks * 32, wr * 128 + mb * 16, l15, l4);
#if VARIANT == 3
      // ablation: keep fragments alive without MFMAs (guide rule 17:
      // plain #if-out would DCE the tr reads and the staging)
#pragma unroll
      for (int mb = 0; mb < 8; ++mb) asm volatile("" :: "v"(afr[mb]));
#pragma unroll
      for (int nb = 0; nb < 4; ++nb) asm volatile("" :: "v"(bfr[nb]));
#else
#pragma unroll
      for (int mb = 0; mb < 8; ++mb)
#pragma unroll
        for (int nb = 0; nb < 4; ++nb)
          acc[mb][nb] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afr[mb], bfr[nb], acc[mb][nb], 0, 0, 0);
#endif
    }
    __builtin_amdgcn_s_setprio(0);

#if VARIANT == 1
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");  // tr reads done
    __builtin_amdgcn_s_barrier();   // raw: does NOT drain vmcnt
    if (t + 2 < nsteps) {           // refill the buffer just freed
      const long long koff = (t + 2) * BK;
      char* nxt = smem + (t & 1) * 65536;
      stage_glds(asrc + koff * lda, lda, nxt);
      stage_glds(bsrc + koff * ldb2, ldb2, nxt + 32768);
      if (t + 1 < nsteps)
        asm volatile("s_waitcnt vmcnt(8)" ::: "memory");  // t+1 landed
    } else if (t + 1 < nsteps) {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");    // nothing queued behind
    }
#else
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
#endif
  }

  // epilogue: C row = l4*4 + r, col = l15 per 16x16 fragment
  const long long mbase = (long long)tm * BM + wr * 128;
  const long long nbase = (long long)tn * BN + wc * 64;
#pragma unroll
  for (int mb = 0; mb < 8; ++mb)
#pragma unroll
    for (int r4 = 0; r4 < 4; ++r4) {
      const long long m = mbase + mb * 16 + l4 * 4 + r4;
#pragma unroll
      for (int nb = 0; nb < 4; ++nb) {
        const long long n = nbase + nb * 16 + l15;
        const float v = ((float*)&acc[mb][nb])[r4];
        if (splits > 1) atomicAdd(dw_acc + m * N + n, v);
        else dw_acc[m * N + n] = v;
      }
    }
}

__global__ void cast_bf16_kernel(const float* __restrict__ src,
                                 short* __restrict__ dst, long long n) {
  long long i = blockIdx.x * 256LL + threadIdx.x;
  if (i < n) dst[i] = f2bf(src[i]);
}

// ---------------------------------------------------------------------------
// tr-mapping self-test: fill a swizzled [64][256] image with id = k*256+m,
// read fragments back, check lane l elem j == (kb + l4*8 + j, m0 + l15).
// ---------------------------------------------------------------------------
__global__ void tr_probe_kernel(short* out) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  for (int i = threadIdx.x; i < 64 * 256; i += blockDim.x) {
    int k = i / 256, m = i % 256;
    *(short*)(smem + k * 512 + ((m * 2) ^ (uk(k) * 32))) =
        (short)(k * 256 + m);
  }
  __syncthreads();
  if (threadIdx.x < 64) {
    int l15 = (int)threadIdx.x & 15, l4 = (int)threadIdx.x >> 4;
    bf16x8 f1 = frag_tr(smem, 0, 16, l15, l4);    // kb=0,  m0=16
    bf16x8 f2 = frag_tr(smem, 32, 240, l15, l4);  // kb=32, m0=240
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      out[threadIdx.x * 16 + j] = ((short*)&f1)[j];
      out[threadIdx.x * 16 + 8 + j] = ((short*)&f2)[j];
    }
  }
}

static int run_tr_probe() {
  short* d;
  hipMalloc(&d, 64 * 16 * 2);
  tr_probe_kernel<<<1, 256, 64 * 512>>>(d);
  hipDeviceSynchronize();
  if (hipGetLastError() != hipSuccess) { printf("tr probe HIP err\n"); return 1; }
  std::vector<short> h(64 * 16);
  hipMemcpy(h.data(), d, 64 * 16 * 2, hipMemcpyDeviceToHost);
  hipFree(d);
  int bad = 0;
  for (int l = 0; l < 64; ++l) {
    int l15 = l & 15, l4 = l >> 4;
    for (int j = 0; j < 8; ++j) {
      short want1 = (short)((0 + l4 * 8 + j) * 256 + 16 + l15);
      short want2 = (short)((32 + l4 * 8 + j) * 256 + 240 + l15);
      if (h[l * 16 + j] != want1 || h[l * 16 + 8 + j] != want2) {
        if (bad < 8)
          printf("tr MISMATCH lane %d j %d: got %d/%d want %d/%d\n", l, j,
                 h[l * 16 + j], h[l * 16 + 8 + j], want1, want2);
        ++bad;
      }
    }
  }
  printf("tr-mapping self-test: %s (%d mismatches)\n", bad ? "FAIL" : "OK", bad);
  return bad ? 1 : 0;
}

int main(int argc, char** argv) {
  int M = argc > 1 ? atoi(argv[1]) : 4608;
  int N = argc > 2 ? atoi(argv[2]) : 1536;
  long long K = argc > 3 ? atoll(argv[3]) : 65536;
  const char* slist = argc > 4 ? argv[4] : "8";
  int iters = argc > 5 ? atoi(argv[5]) : 50;
  if (M % BM || N % BN || K % BK) {
    printf("shape must divide: M%%%d N%%%d K%%%d\n", BM, BN, BK);
    return 1;
  }
  if (run_tr_probe()) return 1;

  std::vector<short> ha((size_t)K * M), hb((size_t)K * N);
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

#if VARIANT == 6 || VARIANT == 7
  const size_t lds = 163840;  // 96 KiB A ring + 64 KiB B pair (full CU LDS)
#else
  const size_t lds = 2 * 65536;  // 128 KiB
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
#if VARIANT == 5
      wgrad2_v5_kernel<<<grid, block, lds>>>(da, db, dacc, M, N, K, S);
#elif VARIANT == 6
      wgrad2_v6_kernel<<<grid, block, lds>>>(da, db, dacc, M, N, K, S);
#elif VARIANT == 7
      wgrad2_v7_kernel<<<grid, block, lds>>>(da, db, dacc, M, N, K, S);
#elif VARIANT == 8
      wgrad2_v8_kernel<<<grid, block, lds>>>(da, db, dacc, M, N, K, S);
#else
      wgrad2_kernel<<<grid, block, lds>>>(da, db, dacc, M, N, K, S);
#endif
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
    printf("G2 VARIANT %d wgrad %dx%dx%lld S=%d: %.1f us  %.1f TF/s\n",
           VARIANT, M, N, K, S, us, tf);

#if VARIANT == 2 || VARIANT == 3
    printf("(ablation variant: results not checked)\n");
    continue;
#endif
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
