// Standalone ablation harness for the local-attention BACKWARD kernel.
// Dev tool (not part of the extension): duplicates the production kernel
// from progen_amd/ops/hip/attention_bwd.hip with structural switches so
// one gpurun call can A/B candidate restructurings (guide rule 24:
// measure within one probe, never across probes). The production kernel
// is 1042 us/call at B=32 H=16 N=1024 wsz=512 (2.9x fwd at 2.5x flops)
// and runs at 154 KiB LDS -> 1 block/CU -> 1 wave/SIMD; the variants
// here probe the obvious levers:
//
//   VARIANT 0: production structure
//   VARIANT 1: no setprio (8 sites)
//   VARIANT 2: atomicAdd dV/dK stores instead of read-modify-write
//              accumulate across chunk rounds (checksum NOT comparable:
//              accumulates across timing iterations)
//   VARIANT 3: no T14 (k/v loads issued at write time, not a tile early)
//   -DNCHUNK=n (default 4): chunks staged per round. NCHUNK=1 shrinks
//              per-chunk LDS 128->32 KiB (58 KiB total -> 2 blocks/CU,
//              2 waves/SIMD) at the cost of 4x the dV/dK global
//              accumulate rounds; NCHUNK=2 is the midpoint (90 KiB).
//
// Build/run (on a GPU box):
//   hipcc --offload-arch=gfx950 -O3 -std=c++17 -DVARIANT=0 \
//       tools/ablate_attn_bwd.hip -o /tmp/abb0 && /tmp/abb0 32 16 1024 512
//
#ifndef VARIANT
#define VARIANT 0
#endif
#ifndef NCHUNK
#define NCHUNK 4
#endif
#include "../progen_amd/ops/hip/common.h"

#define DH 64
#define KT 64
#define ATTN_WAVES 4
#define ATTN_BLOCK (ATTN_WAVES * WAVE)
#define NEG_INF (-1e30f)

__device__ __forceinline__ int swz(int row, int byte_in_row) {
  return (byte_in_row ^ ((row & 7) << 4));
}

__device__ __forceinline__ void load_rope(const float* rsin,
                                          const float* rcos, long long pos,
                                          int d0, float* sv, float* cv) {
  *(f32x4*)(sv) = *(const f32x4*)(rsin + pos * DH + d0);
  *(f32x4*)(sv + 4) = *(const f32x4*)(rsin + pos * DH + d0 + 4);
  *(f32x4*)(cv) = *(const f32x4*)(rcos + pos * DH + d0);
  *(f32x4*)(cv + 4) = *(const f32x4*)(rcos + pos * DH + d0 + 4);
}

__global__ __launch_bounds__(ATTN_BLOCK) void attn_bwd_kernel(
    const short* __restrict__ dout,  // (B, N, H*DH) bf16
    const short* __restrict__ qkv,   // (B, N, 3*H*DH) bf16, PRE-ROTATED
    const short* __restrict__ out,   // (B, N, H*DH) bf16 (fwd output)
    const float* __restrict__ lse,   // (B, H, N)
    float* __restrict__ dacc,        // (B, N, 3*H*DH) fp32 (own + dQ)
    float* __restrict__ dlook,       // (B, N, 2*H*DH) fp32 (lookback k/v)
    int B, int N, int H, int wsz) {
  const int window = blockIdx.x;
  const int head = blockIdx.y;
  const int batch = blockIdx.z;

  const int lane = threadIdx.x % WAVE;
  const int wid = threadIdx.x / WAVE;
  const int l15 = lane & 15;
  const int l4 = lane >> 4;

  const long long HD3 = 3LL * H * DH;
  const long long HD = (long long)H * DH;
  const long long qkv_bn = (long long)batch * N * HD3;
  const long long o_bn = (long long)batch * N * HD;
  const int q_off = head * DH;
  const int k_off = H * DH + head * DH;
  const int v_off = 2 * H * DH + head * DH;
  const long long look_bn = (long long)batch * N * (2LL * H * DH);
  const int lk_off = head * DH;            // k slot in dlook
  const int lv_off = H * DH + head * DH;   // v slot in dlook

  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* k_lds = smem;                                   // 8 KiB
  char* kt_lds = smem + 8192;                           // 8 KiB
  char* v_lds = smem + 16384;                           // 8 KiB
  char* qt_base = smem + 24576;                         // NCHUNK * 8 KiB
  char* dot_base = qt_base + NCHUNK * 8192;
  char* pds_base = dot_base + NCHUNK * 8192;
  char* dsrl_base = pds_base + NCHUNK * 8192;
  float* d_lds = (float*)(dsrl_base + NCHUNK * 8192 + wid * 256);
  float* lse_lds = (float*)(dsrl_base + NCHUNK * 8192 + 1024 + wid * 256);

  char* qt_lds = qt_base + wid * 8192;
  char* dot_lds = dot_base + wid * 8192;
  char* pds_lds = pds_base + wid * 8192;
  char* dsrl_lds = dsrl_base + wid * 8192;

  const float scale = rsqrtf((float)DH);
  const int tiles = 2 * wsz / KT;
  const int chunks = wsz / 64;
  const int rounds = (chunks + NCHUNK - 1) / NCHUNK;

  // T14 staging registers (pure copies of pre-rotated k/v)
  const int su_key[2] = {(int)threadIdx.x >> 3,
                         (int)(threadIdx.x + ATTN_BLOCK) >> 3};
  const int su_d0[2] = {((int)threadIdx.x & 7) * 8,
                        (((int)threadIdx.x + ATTN_BLOCK) & 7) * 8};

  for (int round = 0; round < rounds; ++round) {
    const int chunk = round * NCHUNK + wid;
    const bool active = (wid < NCHUNK) && (chunk < chunks);
    const int nactive = min(NCHUNK, chunks - round * NCHUNK);
    const int chunk_off = chunk * 64;
    const int q0 = window * wsz + chunk_off;

    bf16x8 qfrag[4][2];  // scaled pre-rotated q fragments
    f32x4 dqacc[4][4];
#pragma unroll
    for (int m = 0; m < 4; ++m)
#pragma unroll
      for (int d = 0; d < 4; ++d) dqacc[m][d] = (f32x4){0.f, 0.f, 0.f, 0.f};

    if (active) {
#pragma unroll
      for (int m = 0; m < 4; ++m) {
        const int row = q0 + m * 16 + l15;
#pragma unroll
        for (int ks = 0; ks < 2; ++ks) {
          const int d0 = ks * 32 + 8 * l4;
          bf16x8 v = *(const bf16x8*)(qkv + qkv_bn + (long long)row * HD3 +
                                      q_off + d0);
          bf16x8 o;
#pragma unroll
          for (int j = 0; j < 8; ++j)
            ((short*)&o)[j] = f2bf(bf2f(((short*)&v)[j]) * scale);
          qfrag[m][ks] = o;
        }
      }

      // per-round chunk staging: Q^T (scaled), dO^T, D, lse; one lane/row
      {
        const int row = lane;
        const long long gq = qkv_bn + (long long)(q0 + row) * HD3 + q_off;
        const long long go = o_bn + (long long)(q0 + row) * HD + head * DH;
        float dsum = 0.f;
#pragma unroll
        for (int g = 0; g < 8; ++g) {
          const int d0 = g * 8;
          bf16x8 qv = *(const bf16x8*)(qkv + gq + d0);
          bf16x8 ov = *(const bf16x8*)(out + go + d0);
          bf16x8 dov = *(const bf16x8*)(dout + go + d0);
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            const int d = d0 + j;
            *(short*)(qt_lds + d * 128 + swz(d, row * 2)) =
                f2bf(bf2f(((short*)&qv)[j]) * scale);
            *(short*)(dot_lds + d * 128 + swz(d, row * 2)) = ((short*)&dov)[j];
            dsum += bf2f(((short*)&ov)[j]) * bf2f(((short*)&dov)[j]);
          }
        }
        d_lds[row] = dsum;
        lse_lds[row] = lse[((long long)batch * H + head) * N + q0 + row];
      }
    }
    __syncthreads();  // qt/dot/pds regions ready & previous round done

    const int max_tile = active ? ((chunk_off + 63 + wsz) / KT) : -1;

    // ---- staging prologue (T14): tile 0 ----
    bf16x8 kreg[2], vreg[2];
    auto issue_loads = [&](int t) {
#pragma unroll
      for (int u = 0; u < 2; ++u) {
        const int kpos = (window - 1) * wsz + t * KT + su_key[u];
        if (kpos >= 0) {
          const long long base = qkv_bn + (long long)kpos * HD3;
          kreg[u] = *(const bf16x8*)(qkv + base + k_off + su_d0[u]);
          vreg[u] = *(const bf16x8*)(qkv + base + v_off + su_d0[u]);
        } else {
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            ((short*)&kreg[u])[j] = 0;
            ((short*)&vreg[u])[j] = 0;
          }
        }
      }
    };
    auto write_lds = [&]() {
#pragma unroll
      for (int u = 0; u < 2; ++u) {
        const int key = su_key[u];
        const int d0 = su_d0[u];
        *(bf16x8*)(k_lds + key * 128 + swz(key, d0 * 2)) = kreg[u];
        *(bf16x8*)(v_lds + key * 128 + swz(key, d0 * 2)) = vreg[u];
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const int d = d0 + j;
          *(short*)(kt_lds + d * 128 + swz(d, key * 2)) =
              f2bf(bf2f(((short*)&kreg[u])[j]) * scale);
        }
      }
    };

    issue_loads(0);
    write_lds();
    __syncthreads();
    // (VARIANT 3 still pre-stages tile 0; only steady-state differs)

    for (int t = 0; t < tiles; ++t) {
#if VARIANT != 3
      if (t + 1 < tiles) issue_loads(t + 1);
#endif
      const int kb = t * KT;
      // chunks whose causal range covers this tile: chunk >= c_min
      const int c_min = max(0, (t * KT - wsz) / 64 - round * NCHUNK);
      const bool i_compute = active && t <= max_tile;

      f32x4 s[4][4];  // S -> P for this wave's rows
      if (i_compute) {
#pragma unroll
        for (int m = 0; m < 4; ++m)
#pragma unroll
          for (int n = 0; n < 4; ++n) s[m][n] = (f32x4){0.f, 0.f, 0.f, 0.f};
        #if VARIANT != 1
        __builtin_amdgcn_s_setprio(1);
#endif  // T5: favor MFMA clusters
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
#pragma unroll
          for (int n = 0; n < 4; ++n) {
            const int key = n * 16 + l15;
            bf16x8 kf = *(const bf16x8*)(k_lds + key * 128 +
                                         swz(key, (ks * 32 + 8 * l4) * 2));
#pragma unroll
            for (int m = 0; m < 4; ++m)
              s[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                  qfrag[m][ks], kf, s[m][n], 0, 0, 0);
          }
        #if VARIANT != 1
        __builtin_amdgcn_s_setprio(0);
#endif

        // P = exp(S - lse) masked; b64-write P^T into own pds region
#pragma unroll
        for (int m = 0; m < 4; ++m)
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const int rowiw = chunk_off + m * 16 + l4 * 4 + r;
            const float l = lse_lds[m * 16 + l4 * 4 + r];
#pragma unroll
            for (int n = 0; n < 4; ++n) {
              const int kpos_band = kb + n * 16 + l15;
              float v = ((float*)&s[m][n])[r];
              v = (kpos_band > rowiw + wsz) ? 0.f : __expf(v - l);
              ((float*)&s[m][n])[r] = v;
            }
          }
#pragma unroll
        for (int m = 0; m < 4; ++m)
#pragma unroll
          for (int n = 0; n < 4; ++n) {
            const int key = n * 16 + l15;
            const int row0 = m * 16 + l4 * 4;
            short pk[4];
#pragma unroll
            for (int r = 0; r < 4; ++r) pk[r] = f2bf(((float*)&s[m][n])[r]);
            *(unsigned long long*)(pds_lds + key * 128 + swz(key, row0 * 2)) =
                *(unsigned long long*)pk;
          }
      }
      __syncthreads();  // all P regions ready

      // ---- dV slice: this wave owns keys [wid*16, wid*16+16) of the
      // tile; K-dim spans contributing chunks' rows ----
      {
        f32x4 dv[4];
#pragma unroll
        for (int n = 0; n < 4; ++n) dv[n] = (f32x4){0.f, 0.f, 0.f, 0.f};
        #if VARIANT != 1
        __builtin_amdgcn_s_setprio(1);
#endif
        for (int c = c_min; c < nactive; ++c) {
          char* pds_c = pds_base + c * 8192;
          char* dot_c = dot_base + c * 8192;
#pragma unroll
          for (int ks = 0; ks < 2; ++ks) {
            const int key = wid * 16 + l15;
            const int r0 = ks * 32 + 8 * l4;
            bf16x8 pf = *(const bf16x8*)(pds_c + key * 128 + swz(key, r0 * 2));
#pragma unroll
            for (int n = 0; n < 4; ++n) {
              const int d = n * 16 + l15;
              bf16x8 dof = *(const bf16x8*)(dot_c + d * 128 + swz(d, r0 * 2));
              dv[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pf, dof, dv[n],
                                                              0, 0, 0);
            }
          }
        }
        #if VARIANT != 1
        __builtin_amdgcn_s_setprio(0);
#endif
        const bool lookback = kb < wsz;  // tile-uniform half of the band
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int kpos = (window - 1) * wsz + kb + wid * 16 + l4 * 4 + r;
          if (kpos >= 0) {
            float* dst = lookback
                ? dlook + look_bn + (long long)kpos * (2LL * H * DH) + lv_off
                : dacc + qkv_bn + (long long)kpos * HD3 + v_off;
#pragma unroll
            for (int n = 0; n < 4; ++n) {
              float v = ((float*)&dv[n])[r];
#if VARIANT == 2
              atomicAdd(dst + n * 16 + l15, v);
#else
              if (round > 0) v += dst[n * 16 + l15];  // later chunk rounds
              dst[n * 16 + l15] = v;
#endif
            }
          }
        }
      }

      __syncthreads();  // dV reads of every pds region complete before
                        // any wave overwrites its own with dS

      // ---- dP = dO V'^T ; dS = P o (dP - D); write dS^T + dS ----
      if (i_compute) {
        f32x4 dp[4][4];
#pragma unroll
        for (int m = 0; m < 4; ++m)
#pragma unroll
          for (int n = 0; n < 4; ++n) dp[m][n] = (f32x4){0.f, 0.f, 0.f, 0.f};
        #if VARIANT != 1
        __builtin_amdgcn_s_setprio(1);
#endif
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
#pragma unroll
          for (int m = 0; m < 4; ++m) {
            const int row = q0 + m * 16 + l15;
            const int d0 = ks * 32 + 8 * l4;
            bf16x8 dof = *(const bf16x8*)(dout + o_bn + (long long)row * HD +
                                          head * DH + d0);
#pragma unroll
            for (int n = 0; n < 4; ++n) {
              const int key = n * 16 + l15;
              bf16x8 vf = *(const bf16x8*)(v_lds + key * 128 + swz(key, d0 * 2));
              dp[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                  dof, vf, dp[m][n], 0, 0, 0);
            }
          }
        #if VARIANT != 1
        __builtin_amdgcn_s_setprio(0);
#endif
#pragma unroll
        for (int m = 0; m < 4; ++m)
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const float dval = d_lds[m * 16 + l4 * 4 + r];
#pragma unroll
            for (int n = 0; n < 4; ++n) {
              float p = ((float*)&s[m][n])[r];
              float d = ((float*)&dp[m][n])[r];
              ((float*)&dp[m][n])[r] = p * (d - dval);  // now dS
            }
          }
#pragma unroll
        for (int m = 0; m < 4; ++m)
#pragma unroll
          for (int n = 0; n < 4; ++n) {
            const int key = n * 16 + l15;
            const int row0 = m * 16 + l4 * 4;
            short dk4[4];
#pragma unroll
            for (int r = 0; r < 4; ++r) dk4[r] = f2bf(((float*)&dp[m][n])[r]);
            *(unsigned long long*)(pds_lds + key * 128 + swz(key, row0 * 2)) =
                *(unsigned long long*)dk4;
#pragma unroll
            for (int r = 0; r < 4; ++r)
              *(short*)(dsrl_lds + (row0 + r) * 128 + swz(row0 + r, key * 2)) =
                  dk4[r];
          }
      }
      __syncthreads();  // all dS regions ready

      // ---- dK slice (keys [wid*16, wid*16+16)): K spans chunks ----
      {
        f32x4 dk[4];
#pragma unroll
        for (int n = 0; n < 4; ++n) dk[n] = (f32x4){0.f, 0.f, 0.f, 0.f};
        #if VARIANT != 1
        __builtin_amdgcn_s_setprio(1);
#endif
        for (int c = c_min; c < nactive; ++c) {
          char* pds_c = pds_base + c * 8192;
          char* qt_c = qt_base + c * 8192;
#pragma unroll
          for (int ks = 0; ks < 2; ++ks) {
            const int key = wid * 16 + l15;
            const int r0 = ks * 32 + 8 * l4;
            bf16x8 dsf = *(const bf16x8*)(pds_c + key * 128 + swz(key, r0 * 2));
#pragma unroll
            for (int n = 0; n < 4; ++n) {
              const int d = n * 16 + l15;
              bf16x8 qf = *(const bf16x8*)(qt_c + d * 128 + swz(d, r0 * 2));
              dk[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dsf, qf, dk[n],
                                                              0, 0, 0);
            }
          }
        }
        #if VARIANT != 1
        __builtin_amdgcn_s_setprio(0);
#endif
        const bool lookback = kb < wsz;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int kpos = (window - 1) * wsz + kb + wid * 16 + l4 * 4 + r;
          if (kpos >= 0) {
            float* dst = lookback
                ? dlook + look_bn + (long long)kpos * (2LL * H * DH) + lk_off
                : dacc + qkv_bn + (long long)kpos * HD3 + k_off;
#pragma unroll
            for (int n = 0; n < 4; ++n) {
              float v = ((float*)&dk[n])[r];
#if VARIANT == 2
              atomicAdd(dst + n * 16 + l15, v);
#else
              if (round > 0) v += dst[n * 16 + l15];
              dst[n * 16 + l15] = v;
#endif
            }
          }
        }
      }

      // ---- dQ += dS k_s (own rows; accumulates across tiles) ----
      if (i_compute) {
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
#pragma unroll
          for (int m = 0; m < 4; ++m) {
            const int row = m * 16 + l15;
            bf16x8 dsf = *(const bf16x8*)(dsrl_lds + row * 128 +
                                          swz(row, (ks * 32 + 8 * l4) * 2));
#pragma unroll
            for (int n = 0; n < 4; ++n) {
              const int d = n * 16 + l15;
              bf16x8 kf = *(const bf16x8*)(kt_lds + d * 128 +
                                           swz(d, (ks * 32 + 8 * l4) * 2));
              dqacc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                  dsf, kf, dqacc[m][n], 0, 0, 0);
            }
          }
      }

      __syncthreads();  // done reading k/v/kt LDS for tile t
      if (t + 1 < tiles) {
#if VARIANT == 3
        issue_loads(t + 1);
#endif
        write_lds();
        __syncthreads();
      }
    }

    // ---- store dQ (rows exclusively owned -> plain fp32 stores) ----
    if (active) {
#pragma unroll
      for (int m = 0; m < 4; ++m)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int row = q0 + m * 16 + l4 * 4 + r;
#pragma unroll
          for (int n = 0; n < 4; ++n)
            dacc[qkv_bn + (long long)row * HD3 + q_off + n * 16 + l15] =
                ((float*)&dqacc[m][n])[r];
        }
    }
    __syncthreads();
  }
}

// ---------------------------------------------------------------------------
// VARIANT 4: restructured 2-barrier schedule. The round-2 PMC profile
// at the production grid (B=64 H=24 wsz=256, 6144 blocks) shows
// SQ_WAIT_ANY (barrier/waitcnt-parked) = 50% of wave cycles with only
// 10% issue-stall: the 5-barriers-per-tile phase structure is the
// bottleneck, not the math. This variant:
//   - makes S, P, dP, dS, dQ ALL wave-local in one unbarriered phase
//     (dP does not depend on other waves' P, dS is wave-local, and dQ
//     consumes the wave's OWN dS via ds_read_b64_tr_b16 transposed
//     reads of its freshly written dS^T region — same-wave DS ordering
//     is program order, no barrier);
//   - stores dS^T ONCE (b64 writes into a [key][row] image with the
//     32-B XOR swizzle col ^= u(key)*32, u(k)=(k&3)|((k&8)>>1), which
//     the tr read needs conflict-free) — the old dS row-major scatter
//     image (dsrl, 16 b16 writes per m,n) is GONE;
//   - merges the dV and dK key-slices into one phase (both only read
//     pds/ds2/dot/qt, none of which phase 1 touches after its barrier);
//   - stages tile t+1's k/v/kt DURING phase 2 (phase 2 does not read
//     k/v/kt), removing the separate staging barrier.
// Net: 2 barriers/tile instead of 5. LDS total unchanged (154 KiB:
// dsrl's 32 KiB becomes the dS^T region).
// ---------------------------------------------------------------------------

// per-key 32-B XOR window for the dS^T image (128-B rows): the half-
// wave's 8 key-rows {kb..kb+3, kb+8..kb+11} split 4/4 by parity (row
// base 32*(key&1) dwords) and within a parity v(key) is a bijection
// onto 0..3, so the 8 rows cover all 64 banks -> zero-conflict tr
// reads. v*32 <= 96 B stays inside the 128-B row (uk-style *32 from
// the wgrad image would escape it).
__device__ __forceinline__ int uk4(int k) { return ((k & 2) >> 1) | ((k & 8) >> 2); }

typedef __attribute__((__vector_size__(4 * sizeof(__bf16)))) __bf16 bf16x4t;
#define AS3 __attribute__((address_space(3)))

__global__ __launch_bounds__(ATTN_BLOCK) void attn_bwd_v4_kernel(
    const short* __restrict__ dout, const short* __restrict__ qkv,
    const short* __restrict__ out, const float* __restrict__ lse,
    float* __restrict__ dacc, float* __restrict__ dlook,
    int B, int N, int H, int wsz) {
  const int window = blockIdx.x;
  const int head = blockIdx.y;
  const int batch = blockIdx.z;

  const int lane = threadIdx.x % WAVE;
  const int wid = threadIdx.x / WAVE;
  const int l15 = lane & 15;
  const int l4 = lane >> 4;

  const long long HD3 = 3LL * H * DH;
  const long long HD = (long long)H * DH;
  const long long qkv_bn = (long long)batch * N * HD3;
  const long long o_bn = (long long)batch * N * HD;
  const int q_off = head * DH;
  const int k_off = H * DH + head * DH;
  const int v_off = 2 * H * DH + head * DH;
  const long long look_bn = (long long)batch * N * (2LL * H * DH);
  const int lk_off = head * DH;
  const int lv_off = H * DH + head * DH;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* k_lds = smem;
  char* kt_lds = smem + 8192;
  char* v_lds = smem + 16384;
  char* qt_base = smem + 24576;                  // 32 KiB
  char* dot_base = qt_base + 4 * 8192;           // 32 KiB
  char* pds_base = dot_base + 4 * 8192;          // 32 KiB (P^T)
  char* ds2_base = pds_base + 4 * 8192;          // 32 KiB (dS^T, XOR-u)
  float* d_lds = (float*)(ds2_base + 4 * 8192 + wid * 256);
  float* lse_lds = (float*)(ds2_base + 4 * 8192 + 1024 + wid * 256);

  char* qt_lds = qt_base + wid * 8192;
  char* dot_lds = dot_base + wid * 8192;
  char* pds_lds = pds_base + wid * 8192;
  char* ds2_lds = ds2_base + wid * 8192;

  const float scale = rsqrtf((float)DH);
  const int tiles = 2 * wsz / KT;
  const int chunks = wsz / 64;
  const int rounds = (chunks + 3) / 4;

  const int su_key[2] = {(int)threadIdx.x >> 3,
                         (int)(threadIdx.x + ATTN_BLOCK) >> 3};
  const int su_d0[2] = {((int)threadIdx.x & 7) * 8,
                        (((int)threadIdx.x + ATTN_BLOCK) & 7) * 8};

  for (int round = 0; round < rounds; ++round) {
    const int chunk = round * 4 + wid;
    const bool active = chunk < chunks;
    const int nactive = min(4, chunks - round * 4);
    const int chunk_off = chunk * 64;
    const int q0 = window * wsz + chunk_off;

    bf16x8 qfrag[4][2];
    f32x4 dqacc[4][4];
#pragma unroll
    for (int m = 0; m < 4; ++m)
#pragma unroll
      for (int d = 0; d < 4; ++d) dqacc[m][d] = (f32x4){0.f, 0.f, 0.f, 0.f};

    if (active) {
#pragma unroll
      for (int m = 0; m < 4; ++m) {
        const int row = q0 + m * 16 + l15;
#pragma unroll
        for (int ks = 0; ks < 2; ++ks) {
          const int d0 = ks * 32 + 8 * l4;
          bf16x8 v = *(const bf16x8*)(qkv + qkv_bn + (long long)row * HD3 +
                                      q_off + d0);
          bf16x8 o;
#pragma unroll
          for (int j = 0; j < 8; ++j)
            ((short*)&o)[j] = f2bf(bf2f(((short*)&v)[j]) * scale);
          qfrag[m][ks] = o;
        }
      }
      {
        const int row = lane;
        const long long gq = qkv_bn + (long long)(q0 + row) * HD3 + q_off;
        const long long go = o_bn + (long long)(q0 + row) * HD + head * DH;
        float dsum = 0.f;
#pragma unroll
        for (int g = 0; g < 8; ++g) {
          const int d0 = g * 8;
          bf16x8 qv = *(const bf16x8*)(qkv + gq + d0);
          bf16x8 ov = *(const bf16x8*)(out + go + d0);
          bf16x8 dov = *(const bf16x8*)(dout + go + d0);
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            const int d = d0 + j;
            *(short*)(qt_lds + d * 128 + swz(d, row * 2)) =
                f2bf(bf2f(((short*)&qv)[j]) * scale);
            *(short*)(dot_lds + d * 128 + swz(d, row * 2)) = ((short*)&dov)[j];
            dsum += bf2f(((short*)&ov)[j]) * bf2f(((short*)&dov)[j]);
          }
        }
        d_lds[row] = dsum;
        lse_lds[row] = lse[((long long)batch * H + head) * N + q0 + row];
      }
    }
    __syncthreads();

    const int max_tile = active ? ((chunk_off + 63 + wsz) / KT) : -1;

    bf16x8 kreg[2], vreg[2];
    auto issue_loads = [&](int t) {
#pragma unroll
      for (int u = 0; u < 2; ++u) {
        const int kpos = (window - 1) * wsz + t * KT + su_key[u];
        if (kpos >= 0) {
          const long long base = qkv_bn + (long long)kpos * HD3;
          kreg[u] = *(const bf16x8*)(qkv + base + k_off + su_d0[u]);
          vreg[u] = *(const bf16x8*)(qkv + base + v_off + su_d0[u]);
        } else {
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            ((short*)&kreg[u])[j] = 0;
            ((short*)&vreg[u])[j] = 0;
          }
        }
      }
    };
    auto write_lds = [&]() {
#pragma unroll
      for (int u = 0; u < 2; ++u) {
        const int key = su_key[u];
        const int d0 = su_d0[u];
        *(bf16x8*)(k_lds + key * 128 + swz(key, d0 * 2)) = kreg[u];
        *(bf16x8*)(v_lds + key * 128 + swz(key, d0 * 2)) = vreg[u];
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const int d = d0 + j;
          *(short*)(kt_lds + d * 128 + swz(d, key * 2)) =
              f2bf(bf2f(((short*)&kreg[u])[j]) * scale);
        }
      }
    };

    issue_loads(0);
    write_lds();
    __syncthreads();

    for (int t = 0; t < tiles; ++t) {
      if (t + 1 < tiles) issue_loads(t + 1);
      const int kb = t * KT;
      const int c_min = max(0, (t * KT - wsz) / 64 - round * 4);
      const bool i_compute = active && t <= max_tile;

      // ---- phase 1 (wave-local): S, P, P^T, dP, dS, dS^T, dQ ----
      if (i_compute) {
        f32x4 s[4][4];
#pragma unroll
        for (int m = 0; m < 4; ++m)
#pragma unroll
          for (int n = 0; n < 4; ++n) s[m][n] = (f32x4){0.f, 0.f, 0.f, 0.f};
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
#pragma unroll
          for (int n = 0; n < 4; ++n) {
            const int key = n * 16 + l15;
            bf16x8 kf = *(const bf16x8*)(k_lds + key * 128 +
                                         swz(key, (ks * 32 + 8 * l4) * 2));
#pragma unroll
            for (int m = 0; m < 4; ++m)
              s[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                  qfrag[m][ks], kf, s[m][n], 0, 0, 0);
          }
        __builtin_amdgcn_s_setprio(0);

#pragma unroll
        for (int m = 0; m < 4; ++m)
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const int rowiw = chunk_off + m * 16 + l4 * 4 + r;
            const float l = lse_lds[m * 16 + l4 * 4 + r];
#pragma unroll
            for (int n = 0; n < 4; ++n) {
              const int kpos_band = kb + n * 16 + l15;
              float v = ((float*)&s[m][n])[r];
              v = (kpos_band > rowiw + wsz) ? 0.f : __expf(v - l);
              ((float*)&s[m][n])[r] = v;
            }
          }
#pragma unroll
        for (int m = 0; m < 4; ++m)
#pragma unroll
          for (int n = 0; n < 4; ++n) {
            const int key = n * 16 + l15;
            const int row0 = m * 16 + l4 * 4;
            short pk[4];
#pragma unroll
            for (int r = 0; r < 4; ++r) pk[r] = f2bf(((float*)&s[m][n])[r]);
            *(unsigned long long*)(pds_lds + key * 128 + swz(key, row0 * 2)) =
                *(unsigned long long*)pk;
          }

        // dP = dO V'^T (wave-local: v_lds staged, dout from global)
        f32x4 dp[4][4];
#pragma unroll
        for (int m = 0; m < 4; ++m)
#pragma unroll
          for (int n = 0; n < 4; ++n) dp[m][n] = (f32x4){0.f, 0.f, 0.f, 0.f};
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
#pragma unroll
          for (int m = 0; m < 4; ++m) {
            const int row = q0 + m * 16 + l15;
            const int d0 = ks * 32 + 8 * l4;
            bf16x8 dof = *(const bf16x8*)(dout + o_bn + (long long)row * HD +
                                          head * DH + d0);
#pragma unroll
            for (int n = 0; n < 4; ++n) {
              const int key = n * 16 + l15;
              bf16x8 vf = *(const bf16x8*)(v_lds + key * 128 + swz(key, d0 * 2));
              dp[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                  dof, vf, dp[m][n], 0, 0, 0);
            }
          }
        __builtin_amdgcn_s_setprio(0);
#pragma unroll
        for (int m = 0; m < 4; ++m)
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const float dval = d_lds[m * 16 + l4 * 4 + r];
#pragma unroll
            for (int n = 0; n < 4; ++n) {
              float p = ((float*)&s[m][n])[r];
              float d = ((float*)&dp[m][n])[r];
              ((float*)&dp[m][n])[r] = p * (d - dval);  // now dS
            }
          }
        // write dS^T once: b64 into [key][row] image, col ^= u(key)*32
#pragma unroll
        for (int m = 0; m < 4; ++m)
#pragma unroll
          for (int n = 0; n < 4; ++n) {
            const int key = n * 16 + l15;
            const int row0 = m * 16 + l4 * 4;
            short dk4[4];
#pragma unroll
            for (int r = 0; r < 4; ++r) dk4[r] = f2bf(((float*)&dp[m][n])[r]);
            *(unsigned long long*)(ds2_lds + key * 128 +
                                   ((row0 * 2) ^ (uk4(key) * 32))) =
                *(unsigned long long*)dk4;
          }

        // dQ += dS k_s: A-frags by tr reads of the wave's OWN dS^T
        // (same-wave DS ordering; no barrier needed)
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
#pragma unroll
          for (int m = 0; m < 4; ++m) {
            const int krow1 = ks * 32 + l4 * 8 + (l15 >> 2);
            const int krow2 = krow1 + 4;
            const int colb = (m * 16 + (l15 & 3) * 4) * 2;
            auto p1 = (AS3 bf16x4t*)(ds2_lds + krow1 * 128 +
                                     (colb ^ (uk4(krow1) * 32)));
            auto p2 = (AS3 bf16x4t*)(ds2_lds + krow2 * 128 +
                                     (colb ^ (uk4(krow2) * 32)));
            bf16x4t f1 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p1);
            bf16x4t f2 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p2);
            bf16x8 dsf;
#pragma unroll
            for (int j = 0; j < 4; ++j) {
              ((__bf16*)&dsf)[j] = f1[j];
              ((__bf16*)&dsf)[j + 4] = f2[j];
            }
#pragma unroll
            for (int n = 0; n < 4; ++n) {
              const int d = n * 16 + l15;
              bf16x8 kf = *(const bf16x8*)(kt_lds + d * 128 +
                                           swz(d, (ks * 32 + 8 * l4) * 2));
              dqacc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                  dsf, kf, dqacc[m][n], 0, 0, 0);
            }
          }
        __builtin_amdgcn_s_setprio(0);
      }
      __syncthreads();  // barrier A: P^T/dS^T ready; k/v/kt reads done

      // stage t+1 into k/v/kt while phase 2 runs (phase 2 reads none)
      if (t + 1 < tiles) write_lds();

      // ---- phase 2: merged dV + dK key-slices ----
      {
        f32x4 dv[4], dk[4];
#pragma unroll
        for (int n = 0; n < 4; ++n) {
          dv[n] = (f32x4){0.f, 0.f, 0.f, 0.f};
          dk[n] = (f32x4){0.f, 0.f, 0.f, 0.f};
        }
        __builtin_amdgcn_s_setprio(1);
        for (int c = c_min; c < nactive; ++c) {
          char* pds_c = pds_base + c * 8192;
          char* ds2_c = ds2_base + c * 8192;
          char* dot_c = dot_base + c * 8192;
          char* qt_c = qt_base + c * 8192;
#pragma unroll
          for (int ks = 0; ks < 2; ++ks) {
            const int key = wid * 16 + l15;
            const int r0 = ks * 32 + 8 * l4;
            bf16x8 pf = *(const bf16x8*)(pds_c + key * 128 + swz(key, r0 * 2));
            bf16x8 dsf = *(const bf16x8*)(ds2_c + key * 128 +
                                          ((r0 * 2) ^ (uk4(key) * 32)));
#pragma unroll
            for (int n = 0; n < 4; ++n) {
              const int d = n * 16 + l15;
              bf16x8 dof = *(const bf16x8*)(dot_c + d * 128 + swz(d, r0 * 2));
              bf16x8 qf = *(const bf16x8*)(qt_c + d * 128 + swz(d, r0 * 2));
              dv[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pf, dof, dv[n],
                                                              0, 0, 0);
              dk[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dsf, qf, dk[n],
                                                              0, 0, 0);
            }
          }
        }
        __builtin_amdgcn_s_setprio(0);
        const bool lookback = kb < wsz;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int kpos = (window - 1) * wsz + kb + wid * 16 + l4 * 4 + r;
          if (kpos >= 0) {
            float* dstv = lookback
                ? dlook + look_bn + (long long)kpos * (2LL * H * DH) + lv_off
                : dacc + qkv_bn + (long long)kpos * HD3 + v_off;
            float* dstk = lookback
                ? dlook + look_bn + (long long)kpos * (2LL * H * DH) + lk_off
                : dacc + qkv_bn + (long long)kpos * HD3 + k_off;
#pragma unroll
            for (int n = 0; n < 4; ++n) {
              float vv = ((float*)&dv[n])[r];
              float vk = ((float*)&dk[n])[r];
              if (round > 0) {
                vv += dstv[n * 16 + l15];
                vk += dstk[n * 16 + l15];
              }
              dstv[n * 16 + l15] = vv;
              dstk[n * 16 + l15] = vk;
            }
          }
        }
      }
      __syncthreads();  // barrier B: slices done; t+1 staged
    }

    if (active) {
#pragma unroll
      for (int m = 0; m < 4; ++m)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int row = q0 + m * 16 + l4 * 4 + r;
#pragma unroll
          for (int n = 0; n < 4; ++n)
            dacc[qkv_bn + (long long)row * HD3 + q_off + n * 16 + l15] =
                ((float*)&dqacc[m][n])[r];
        }
    }
    __syncthreads();
  }
}


// ---------------------------------------------------------------------------
// VARIANT 5: 8-wave geometry — 2 waves/SIMD. V4's remaining PMC profile
// is 47.8% barrier/wait-parked at ONE wave/SIMD (154 KiB LDS = 1 block
// /CU, 501 VGPRs = 1 wave/SIMD): nothing covers a parked wave. Here a
// block is 8 waves of 32-row chunks (per-wave registers halve: s/dp/
// dqacc are [2][4]), so each SIMD holds TWO waves that cover each
// other's stalls. Wave pairs share the 64-row chunk regions (qt/dot/
// pds/ds2 keep their 128-B-row layouts and swizzles); the dV/dK key
// slices split the MFMA K-dim by ks-half across wave halves, the upper
// half writing to separate dacc2/dlook2 accumulators (plain stores
// stay raceless; the finalize pass sums four sources instead of two).
// ---------------------------------------------------------------------------

#define V5_WAVES 8
#define V5_BLOCK (V5_WAVES * WAVE)

__global__ __launch_bounds__(V5_BLOCK) void attn_bwd_v5_kernel(
    const short* __restrict__ dout, const short* __restrict__ qkv,
    const short* __restrict__ out, const float* __restrict__ lse,
    float* __restrict__ dacc, float* __restrict__ dlook,
    float* __restrict__ dacc2, float* __restrict__ dlook2,
    int B, int N, int H, int wsz) {
  const int window = blockIdx.x;
  const int head = blockIdx.y;
  const int batch = blockIdx.z;

  const int lane = threadIdx.x % WAVE;
  const int wid = threadIdx.x / WAVE;   // 0..7
  const int l15 = lane & 15;
  const int l4 = lane >> 4;
  const int pid = wid >> 1;             // pair region 0..3
  const int phalf = wid & 1;            // row half within the pair

  const long long HD3 = 3LL * H * DH;
  const long long HD = (long long)H * DH;
  const long long HD2 = 2LL * H * DH;
  const long long qkv_bn = (long long)batch * N * HD3;
  const long long o_bn = (long long)batch * N * HD;
  const int q_off = head * DH;
  const int k_off = H * DH + head * DH;
  const int v_off = 2 * H * DH + head * DH;
  const long long look_bn = (long long)batch * N * HD2;
  const int lk_off = head * DH;
  const int lv_off = H * DH + head * DH;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* k_lds = smem;
  char* kt_lds = smem + 8192;
  char* v_lds = smem + 16384;
  char* qt_base = smem + 24576;
  char* dot_base = qt_base + 4 * 8192;
  char* pds_base = dot_base + 4 * 8192;
  char* ds2_base = pds_base + 4 * 8192;
  float* d_base = (float*)(ds2_base + 4 * 8192);        // [pair][64]
  float* lse_base = (float*)(ds2_base + 4 * 8192 + 1024);

  char* qt_lds = qt_base + pid * 8192;
  char* dot_lds = dot_base + pid * 8192;
  char* pds_lds = pds_base + pid * 8192;
  char* ds2_lds = ds2_base + pid * 8192;
  float* d_lds = d_base + pid * 64;
  float* lse_lds = lse_base + pid * 64;

  const float scale = rsqrtf((float)DH);
  const int tiles = 2 * wsz / KT;
  const int chunks64 = wsz / 64;                 // pair-sized chunks
  const int rounds = (chunks64 + 3) / 4;

  // block-wide k/v tile staging: 512 threads cover 64 keys x 64 dh
  const int su_key = (int)threadIdx.x >> 3;
  const int su_d0 = ((int)threadIdx.x & 7) * 8;

  for (int round = 0; round < rounds; ++round) {
    const int c64 = round * 4 + pid;             // this pair's 64-chunk
    const bool active = c64 < chunks64;
    const int nactive = min(4, chunks64 - round * 4);
    const int chunk_off = c64 * 64 + phalf * 32; // rows-in-window
    const int q0 = window * wsz + chunk_off;
    const int colbase = phalf * 32;              // in the pair region

    bf16x8 qfrag[2][2];
    f32x4 dqacc[2][4];
#pragma unroll
    for (int m = 0; m < 2; ++m)
#pragma unroll
      for (int d = 0; d < 4; ++d) dqacc[m][d] = (f32x4){0.f, 0.f, 0.f, 0.f};

    if (active) {
#pragma unroll
      for (int m = 0; m < 2; ++m) {
        const int row = q0 + m * 16 + l15;
#pragma unroll
        for (int ks = 0; ks < 2; ++ks) {
          const int d0 = ks * 32 + 8 * l4;
          bf16x8 v = *(const bf16x8*)(qkv + qkv_bn + (long long)row * HD3 +
                                      q_off + d0);
          bf16x8 o;
#pragma unroll
          for (int j = 0; j < 8; ++j)
            ((short*)&o)[j] = f2bf(bf2f(((short*)&v)[j]) * scale);
          qfrag[m][ks] = o;
        }
      }
      // stage this wave's 32 rows into the pair region columns
      // [colbase, colbase+32): 2 lanes per row, 4 d-groups each
      {
        const int row = lane >> 1;
        const int col = colbase + row;
        const long long gq = qkv_bn + (long long)(q0 + row) * HD3 + q_off;
        const long long go = o_bn + (long long)(q0 + row) * HD + head * DH;
        float dsum = 0.f;
#pragma unroll
        for (int gg = 0; gg < 4; ++gg) {
          const int g = (lane & 1) * 4 + gg;
          const int d0 = g * 8;
          bf16x8 qv = *(const bf16x8*)(qkv + gq + d0);
          bf16x8 ov = *(const bf16x8*)(out + go + d0);
          bf16x8 dov = *(const bf16x8*)(dout + go + d0);
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            const int d = d0 + j;
            *(short*)(qt_lds + d * 128 + swz(d, col * 2)) =
                f2bf(bf2f(((short*)&qv)[j]) * scale);
            *(short*)(dot_lds + d * 128 + swz(d, col * 2)) = ((short*)&dov)[j];
            dsum += bf2f(((short*)&ov)[j]) * bf2f(((short*)&dov)[j]);
          }
        }
        dsum += __shfl_xor(dsum, 1, 64);  // join the row's two lanes
        if ((lane & 1) == 0) {
          d_lds[col] = dsum;
          lse_lds[col] = lse[((long long)batch * H + head) * N + q0 + row];
        }
      }
    }
    __syncthreads();

    const int max_tile = active ? ((chunk_off + 31 + wsz) / KT) : -1;

    bf16x8 kreg, vreg;
    auto issue_loads = [&](int t) {
      const int kpos = (window - 1) * wsz + t * KT + su_key;
      if (kpos >= 0) {
        const long long base = qkv_bn + (long long)kpos * HD3;
        kreg = *(const bf16x8*)(qkv + base + k_off + su_d0);
        vreg = *(const bf16x8*)(qkv + base + v_off + su_d0);
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          ((short*)&kreg)[j] = 0;
          ((short*)&vreg)[j] = 0;
        }
      }
    };
    auto write_lds = [&]() {
      *(bf16x8*)(k_lds + su_key * 128 + swz(su_key, su_d0 * 2)) = kreg;
      *(bf16x8*)(v_lds + su_key * 128 + swz(su_key, su_d0 * 2)) = vreg;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int d = su_d0 + j;
        *(short*)(kt_lds + d * 128 + swz(d, su_key * 2)) =
            f2bf(bf2f(((short*)&kreg)[j]) * scale);
      }
    };

    issue_loads(0);
    write_lds();
    __syncthreads();

    for (int t = 0; t < tiles; ++t) {
      if (t + 1 < tiles) issue_loads(t + 1);
      const int kb = t * KT;
      const int c_min = max(0, (t * KT - wsz) / 64 - round * 4);
      const bool i_compute = active && t <= max_tile;

      // ---- phase 1 (wave-local) ----
      if (i_compute) {
        f32x4 s[2][4];
#pragma unroll
        for (int m = 0; m < 2; ++m)
#pragma unroll
          for (int n = 0; n < 4; ++n) s[m][n] = (f32x4){0.f, 0.f, 0.f, 0.f};
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
#pragma unroll
          for (int n = 0; n < 4; ++n) {
            const int key = n * 16 + l15;
            bf16x8 kf = *(const bf16x8*)(k_lds + key * 128 +
                                         swz(key, (ks * 32 + 8 * l4) * 2));
#pragma unroll
            for (int m = 0; m < 2; ++m)
              s[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                  qfrag[m][ks], kf, s[m][n], 0, 0, 0);
          }
        __builtin_amdgcn_s_setprio(0);

#pragma unroll
        for (int m = 0; m < 2; ++m)
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const int rowiw = chunk_off + m * 16 + l4 * 4 + r;
            const float l = lse_lds[colbase + m * 16 + l4 * 4 + r];
#pragma unroll
            for (int n = 0; n < 4; ++n) {
              const int kpos_band = kb + n * 16 + l15;
              float v = ((float*)&s[m][n])[r];
              v = (kpos_band > rowiw + wsz) ? 0.f : __expf(v - l);
              ((float*)&s[m][n])[r] = v;
            }
          }
#pragma unroll
        for (int m = 0; m < 2; ++m)
#pragma unroll
          for (int n = 0; n < 4; ++n) {
            const int key = n * 16 + l15;
            const int row0 = colbase + m * 16 + l4 * 4;
            short pk[4];
#pragma unroll
            for (int r = 0; r < 4; ++r) pk[r] = f2bf(((float*)&s[m][n])[r]);
            *(unsigned long long*)(pds_lds + key * 128 + swz(key, row0 * 2)) =
                *(unsigned long long*)pk;
          }

        f32x4 dp[2][4];
#pragma unroll
        for (int m = 0; m < 2; ++m)
#pragma unroll
          for (int n = 0; n < 4; ++n) dp[m][n] = (f32x4){0.f, 0.f, 0.f, 0.f};
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
#pragma unroll
          for (int m = 0; m < 2; ++m) {
            const int row = q0 + m * 16 + l15;
            const int d0 = ks * 32 + 8 * l4;
            bf16x8 dof = *(const bf16x8*)(dout + o_bn + (long long)row * HD +
                                          head * DH + d0);
#pragma unroll
            for (int n = 0; n < 4; ++n) {
              const int key = n * 16 + l15;
              bf16x8 vf = *(const bf16x8*)(v_lds + key * 128 + swz(key, d0 * 2));
              dp[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                  dof, vf, dp[m][n], 0, 0, 0);
            }
          }
        __builtin_amdgcn_s_setprio(0);
#pragma unroll
        for (int m = 0; m < 2; ++m)
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const float dval = d_lds[colbase + m * 16 + l4 * 4 + r];
#pragma unroll
            for (int n = 0; n < 4; ++n) {
              float p = ((float*)&s[m][n])[r];
              float d = ((float*)&dp[m][n])[r];
              ((float*)&dp[m][n])[r] = p * (d - dval);
            }
          }
#pragma unroll
        for (int m = 0; m < 2; ++m)
#pragma unroll
          for (int n = 0; n < 4; ++n) {
            const int key = n * 16 + l15;
            const int row0 = colbase + m * 16 + l4 * 4;
            short dk4[4];
#pragma unroll
            for (int r = 0; r < 4; ++r) dk4[r] = f2bf(((float*)&dp[m][n])[r]);
            *(unsigned long long*)(ds2_lds + key * 128 +
                                   ((row0 * 2) ^ (uk4(key) * 32))) =
                *(unsigned long long*)dk4;
          }

        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
#pragma unroll
          for (int m = 0; m < 2; ++m) {
            const int krow1 = ks * 32 + l4 * 8 + (l15 >> 2);
            const int krow2 = krow1 + 4;
            const int colb = (colbase + m * 16 + (l15 & 3) * 4) * 2;
            auto p1 = (AS3 bf16x4t*)(ds2_lds + krow1 * 128 +
                                     (colb ^ (uk4(krow1) * 32)));
            auto p2 = (AS3 bf16x4t*)(ds2_lds + krow2 * 128 +
                                     (colb ^ (uk4(krow2) * 32)));
            bf16x4t f1 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p1);
            bf16x4t f2 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p2);
            bf16x8 dsf;
#pragma unroll
            for (int j = 0; j < 4; ++j) {
              ((__bf16*)&dsf)[j] = f1[j];
              ((__bf16*)&dsf)[j + 4] = f2[j];
            }
#pragma unroll
            for (int n = 0; n < 4; ++n) {
              const int d = n * 16 + l15;
              bf16x8 kf = *(const bf16x8*)(kt_lds + d * 128 +
                                           swz(d, (ks * 32 + 8 * l4) * 2));
              dqacc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                  dsf, kf, dqacc[m][n], 0, 0, 0);
            }
          }
        __builtin_amdgcn_s_setprio(0);
      }
      __syncthreads();  // barrier A

      if (t + 1 < tiles) write_lds();

      // ---- phase 2: dV+dK key slices, K-dim split by ks across wave
      // halves (half 1 accumulates into dacc2/dlook2) ----
      {
        const int ks = wid >> 2;          // fixed ks half per wave
        const int keyslot = (wid & 3) * 16 + l15;
        f32x4 dv[4], dk[4];
#pragma unroll
        for (int n = 0; n < 4; ++n) {
          dv[n] = (f32x4){0.f, 0.f, 0.f, 0.f};
          dk[n] = (f32x4){0.f, 0.f, 0.f, 0.f};
        }
        __builtin_amdgcn_s_setprio(1);
        const int r0 = ks * 32 + 8 * l4;
        for (int c = c_min; c < nactive; ++c) {
          char* pds_c = pds_base + c * 8192;
          char* ds2_c = ds2_base + c * 8192;
          char* dot_c = dot_base + c * 8192;
          char* qt_c = qt_base + c * 8192;
          bf16x8 pf = *(const bf16x8*)(pds_c + keyslot * 128 +
                                       swz(keyslot, r0 * 2));
          bf16x8 dsf = *(const bf16x8*)(ds2_c + keyslot * 128 +
                                        ((r0 * 2) ^ (uk4(keyslot) * 32)));
#pragma unroll
          for (int n = 0; n < 4; ++n) {
            const int d = n * 16 + l15;
            bf16x8 dof = *(const bf16x8*)(dot_c + d * 128 + swz(d, r0 * 2));
            bf16x8 qf = *(const bf16x8*)(qt_c + d * 128 + swz(d, r0 * 2));
            dv[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pf, dof, dv[n],
                                                            0, 0, 0);
            dk[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dsf, qf, dk[n],
                                                            0, 0, 0);
          }
        }
        __builtin_amdgcn_s_setprio(0);
        __syncthreads();  // B1: slices done reading pds/ds2 everywhere
        // half 1 parks its partial in the (now dead) ds2 region scratch:
        // [slot = key-slice][key16 x dh64 f32] x2 (dv, dk)
        float* scratch = (float*)(ds2_base) + (wid & 3) * 2048;
        if (ks == 1) {
#pragma unroll
          for (int n = 0; n < 4; ++n)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
              scratch[(l4 * 4 + r) * 64 + n * 16 + l15] = ((float*)&dv[n])[r];
              scratch[1024 + (l4 * 4 + r) * 64 + n * 16 + l15] =
                  ((float*)&dk[n])[r];
            }
        }
        __syncthreads();  // B2: partials visible
        if (ks == 0) {
          const bool lookback = kb < wsz;
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const int kpos = (window - 1) * wsz + kb + (wid & 3) * 16 + l4 * 4 + r;
            if (kpos >= 0) {
              float* dstv = lookback
                  ? dlook + look_bn + (long long)kpos * HD2 + lv_off
                  : dacc + qkv_bn + (long long)kpos * HD3 + v_off;
              float* dstk = lookback
                  ? dlook + look_bn + (long long)kpos * HD2 + lk_off
                  : dacc + qkv_bn + (long long)kpos * HD3 + k_off;
#pragma unroll
              for (int n = 0; n < 4; ++n) {
                float vv = ((float*)&dv[n])[r] +
                           scratch[(l4 * 4 + r) * 64 + n * 16 + l15];
                float vk = ((float*)&dk[n])[r] +
                           scratch[1024 + (l4 * 4 + r) * 64 + n * 16 + l15];
                if (round > 0) {
                  vv += dstv[n * 16 + l15];
                  vk += dstk[n * 16 + l15];
                }
                dstv[n * 16 + l15] = vv;
                dstk[n * 16 + l15] = vk;
              }
            }
          }
        }
      }
      __syncthreads();  // barrier B3: scratch reads done before t+1's
                        // phase-1 overwrites ds2
    }

    if (active) {
#pragma unroll
      for (int m = 0; m < 2; ++m)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int row = q0 + m * 16 + l4 * 4 + r;
#pragma unroll
          for (int n = 0; n < 4; ++n)
            dacc[qkv_bn + (long long)row * HD3 + q_off + n * 16 + l15] =
                ((float*)&dqacc[m][n])[r];
        }
    }
    __syncthreads();
  }
}

// ---------------------------------------------------------------------------
// finalize: inverse rotary rotation on the fp32 accumulator -> bf16 dqkv
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(256) void attn_bwd_finalize_kernel(
    const float* __restrict__ dacc, const float* __restrict__ dlook,
    const float* __restrict__ rsin, const float* __restrict__ rcos,
    short* __restrict__ dqkv, int B, int N, int H, int wsz) {
  const long long HD3 = 3LL * H * DH;
  const long long HD2 = 2LL * H * DH;
  const long long total = (long long)B * N * 3 * H * (DH / 8);
  for (long long idx = blockIdx.x * 256LL + threadIdx.x; idx < total;
       idx += (long long)gridDim.x * 256) {
    const int g = idx % (DH / 8);
    const long long rest = idx / (DH / 8);
    const int hslot = rest % (3 * H);
    const long long bn = rest / (3 * H);
    const int n = bn % N;
    const int d0 = g * 8;

    const long long off = bn * HD3 + (long long)hslot * DH + d0;
    float x[8], sv[8], cv[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) x[j] = dacc[off + j];
    // k/v slots: add the lookback contribution (exists unless this is
    // the last window — its keys are nobody's lookback)
    if (hslot >= H && (n / wsz) < (N / wsz) - 1) {
      const long long loff = bn * HD2 + (long long)(hslot - H) * DH + d0;
#pragma unroll
      for (int j = 0; j < 8; ++j) x[j] += dlook[loff + j];
    }
    load_rope(rsin, rcos, n, d0, sv, cv);
    // inverse rotation: dx[2i] = dy[2i] c + dy[2i+1] s;
    //                   dx[2i+1] = dy[2i+1] c - dy[2i] s
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      float y0 = x[2 * p], y1 = x[2 * p + 1];
      float s = sv[2 * p], c = cv[2 * p];
      x[2 * p] = y0 * c + y1 * s;
      x[2 * p + 1] = y1 * c - y0 * s;
    }
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) ((short*)&o)[j] = f2bf(x[j]);
    *(bf16x8*)(dqkv + off) = o;
  }
}


#include <cstdio>
#include <cstdlib>
#include <cmath>
#include <vector>

int main(int argc, char** argv) {
  int B = argc > 1 ? atoi(argv[1]) : 32;
  int H = argc > 2 ? atoi(argv[2]) : 16;
  int N = argc > 3 ? atoi(argv[3]) : 1024;
  int wsz = argc > 4 ? atoi(argv[4]) : 512;
  int iters = argc > 5 ? atoi(argv[5]) : 100;

  long long qn = (long long)B * N * 3 * H * DH;
  long long on = (long long)B * N * H * DH;
  long long ln = (long long)B * H * N;
  std::vector<short> hq(qn), ho(on), hdo(on);
  std::vector<float> hl(ln);
  srand(11);
  auto rb = []() {
    float f = ((float)rand() / RAND_MAX - 0.5f) * 0.25f;
    union { float f; unsigned u; } c; c.f = f;
    return (short)(c.u >> 16);
  };
  for (long long i = 0; i < qn; ++i) hq[i] = rb();
  for (long long i = 0; i < on; ++i) { ho[i] = rb(); hdo[i] = rb(); }
  // plausible lse: softmax denominators over the 2*wsz band of tiny
  // logits land near log(2*wsz); keeps P = exp(S - lse) in (0, ~1)
  for (long long i = 0; i < ln; ++i) hl[i] = logf(2.0f * wsz);

  short *dq, *ddo, *dou;
  float *dl, *dacc, *dlook;
  hipMalloc(&dq, qn * 2);
  hipMalloc(&ddo, on * 2);
  hipMalloc(&dou, on * 2);
  hipMalloc(&dl, ln * 4);
  hipMalloc(&dacc, qn * 4);
  hipMalloc(&dlook, (long long)B * N * 2 * H * DH * 4);
  float *dacc2 = nullptr, *dlook2 = nullptr;
#if VARIANT == 5
  hipMalloc(&dacc2, (long long)B * N * 2 * H * DH * 4);
  hipMalloc(&dlook2, (long long)B * N * 2 * H * DH * 4);
  hipMemset(dacc2, 0, (long long)B * N * 2 * H * DH * 4);
  hipMemset(dlook2, 0, (long long)B * N * 2 * H * DH * 4);
#endif
  (void)dacc2; (void)dlook2;
  hipMemcpy(dq, hq.data(), qn * 2, hipMemcpyHostToDevice);
  hipMemcpy(ddo, hdo.data(), on * 2, hipMemcpyHostToDevice);
  hipMemcpy(dou, ho.data(), on * 2, hipMemcpyHostToDevice);
  hipMemcpy(dl, hl.data(), ln * 4, hipMemcpyHostToDevice);
  hipMemset(dacc, 0, qn * 4);
  hipMemset(dlook, 0, (long long)B * N * 2 * H * DH * 4);

  dim3 grid(N / wsz, H, B), block(ATTN_BLOCK);
  size_t lds = 24576 + (size_t)NCHUNK * 4 * 8192 + 2048;
  printf("LDS %zu KiB\n", lds / 1024);

  for (int i = 0; i < 10; ++i)
#if VARIANT == 5
    attn_bwd_v5_kernel<<<grid, dim3(V5_BLOCK), lds>>>(ddo, dq, dou, dl,
                                                      dacc, dlook, dacc2,
                                                      dlook2,
#elif VARIANT == 4
    attn_bwd_v4_kernel<<<grid, block, lds>>>(ddo, dq, dou, dl, dacc, dlook,
#else
    attn_bwd_kernel<<<grid, block, lds>>>(ddo, dq, dou, dl, dacc, dlook,
#endif
                                          B, N, H, wsz);
  hipDeviceSynchronize();
  hipError_t err = hipGetLastError();
  if (err != hipSuccess) { printf("HIP ERR %s\n", hipGetErrorString(err)); return 1; }

  hipEvent_t e0, e1;
  hipEventCreate(&e0);
  hipEventCreate(&e1);
  hipEventRecord(e0);
  for (int i = 0; i < iters; ++i)
#if VARIANT == 5
    attn_bwd_v5_kernel<<<grid, dim3(V5_BLOCK), lds>>>(ddo, dq, dou, dl,
                                                      dacc, dlook, dacc2,
                                                      dlook2,
#elif VARIANT == 4
    attn_bwd_v4_kernel<<<grid, block, lds>>>(ddo, dq, dou, dl, dacc, dlook,
#else
    attn_bwd_kernel<<<grid, block, lds>>>(ddo, dq, dou, dl, dacc, dlook,
#endif
                                          B, N, H, wsz);
  hipEventRecord(e1);
  hipEventSynchronize(e1);
  float ms;
  hipEventElapsedTime(&ms, e0, e1);
  double us = ms * 1000.0 / iters;
  // 5 banded GEMMs (S, dV, dP, dQ, dK) over the 2*wsz band
  double fl = 5.0 * 2.0 * B * H * (double)N * (2.0 * wsz) * DH;
  printf("VARIANT %d NCHUNK %d: %.1f us/call  %.1f TF/s  (B=%d H=%d N=%d wsz=%d)\n",
         VARIANT, NCHUNK, us, fl / (us * 1e-6) / 1e12, B, N == 0 ? 0 : H, N, wsz);

  std::vector<float> hacc(1 << 20);
  hipMemcpy(hacc.data(), dacc, (size_t)(1 << 20) * 4, hipMemcpyDeviceToHost);
  double cs = 0;
  for (int i = 0; i < (1 << 20); i += 97) cs += hacc[i];
  printf("checksum %.6f (V2/atomic accumulates across iters: not comparable)\n", cs);
  return 0;
}
