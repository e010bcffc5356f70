// Fused local-window attention, backward.
//
// Recompute-based flash backward of ops/hip/attention_fwd.hip (the
// reference differentiates progen.py:83-103 through XLA; here the math
// is explicit):
//   D_i = rowsum(dO_i * O_i)
//   P   = exp(S_masked - lse)          (S recomputed from scaled q', k')
//   dV' = P^T dO
//   dP  = dO V'^T
//   dS  = P o (dP - D)
//   dQ' = dS k_s                       (scale folded into staged K^T)
//   dK' = dS^T q_s                     (scale folded into staged Q^T)
// followed by attn_bwd_finalize_kernel, which applies the inverse rotary
// rotation (rotary is linear, so it commutes with accumulation) and
// casts the fp32 accumulator to bf16 dqkv. Input qkv is PRE-ROTATED
// (ops/hip/rope_qkv.hip).
//
// Geometry: block = one (batch, head, window), 4 waves, each wave owns a
// 64-row q-chunk. Round-2 restructure (harness VARIANT 4,
// tools/ablate_attn_bwd.hip — 1965 -> 1500 us/call at the production
// grid B=64 H=24 wsz=256, bitwise-identical output; PMC showed the old
// 5-barriers-per-tile schedule parked waves on barriers 50% of cycles):
// per 64-key tile there are TWO barriers —
//   phase 1 (wave-local, no barrier): S, P (P^T b64-written), dP,
//     dS = P o (dP - D) (dS^T b64-written), and dQ += dS k_s where the
//     dS A-fragments come from ds_read_b64_tr_b16 transposed reads of
//     the wave's OWN just-written dS^T region (same-wave DS ordering is
//     program order). The old separate row-major dS image (dsrl, 16
//     scattered b16 writes per m,n) is gone.
//   barrier A; then tile t+1's k/v/kt staging (phase 2 reads none of
//     them) overlaps with phase 2: the dV and dK 16-key OUTPUT SLICES,
//     merged into one chunk-loop, whose MFMA K-dim spans ALL chunks'
//     P^T/dS^T/dO^T/Q^T regions; barrier B.
// Every dV/dK element is produced by exactly ONE wave, so the stores
// are PLAIN (no atomics): each window's own-band gradients go to dacc,
// its lookback-band gradients to a separate dlook buffer, and
// attn_bwd_finalize_kernel sums the two (adjacent windows share keys
// through the lookback, progen.py:90-91). dQ rows are exclusively
// owned -> plain fp32 stores. Window 0's lookback keys are the zero
// pad; their gradients are discarded.
//
// MFMA operand LDS images (XOR-swizzled, byte ^= (row&7)<<4 except ds2):
//   k_lds  [key][dh]   k'           (S B-fragments)
//   kt_lds [dh][key]   scaled k'    (dQ B-fragments)
//   v_lds  [key][dh]   v'           (dP B-fragments)
//   qt_lds [dh][row]   scaled q'    (dK B-fragments, per chunk)
//   dot_lds[dh][row]   dO           (dV B-fragments, per chunk)
//   pds_lds [key][row] P^T          (b64-written from the MFMA C-layout;
//                                    per chunk)
//   ds2_lds [key][row] dS^T         (b64-written, col ^= v(key)*32 with
//                                    v(k)=((k&2)>>1)|((k&8)>>2) so the
//                                    tr reads land conflict-free; dK
//                                    A-frags read it b128, dQ A-frags
//                                    via tr; per chunk)

#include "common.h"

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

// per-key 32-B XOR window for the dS^T image (128-B rows): the half-
// wave's 8 key-rows {kb..kb+3, kb+8..kb+11} split 4/4 by parity (row
// base 32*(key&1) dwords) and within a parity v(key) is a bijection
// onto 0..3, so the 8 rows cover all 64 banks -> zero-conflict tr
// reads. v*32 <= 96 B stays inside the 128-B row (uk-style *32 from
// the wgrad image would escape it).
__device__ __forceinline__ int uk4(int k) { return ((k & 2) >> 1) | ((k & 8) >> 2); }

typedef __attribute__((__vector_size__(4 * sizeof(__bf16)))) __bf16 bf16x4t;
#define AS3 __attribute__((address_space(3)))

__global__ __launch_bounds__(ATTN_BLOCK) void attn_bwd_kernel(
    const short* __restrict__ dout, const short* __restrict__ qkv,
    const short* __restrict__ halo,   // CP lookback for window 0 (see
                                      // attention_fwd.hip) or nullptr
    const short* __restrict__ out, const float* __restrict__ lse,
    float* __restrict__ dacc, float* __restrict__ dlook,
    float* __restrict__ dhalo,        // (B, wsz, 2*H*DH) fp32 window-0
                                      // lookback grads (CP) or nullptr
                                      // (quirk path: grads discarded)
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
        } else if (halo != nullptr) {
          const long long hb =
              ((long long)batch * wsz + (kpos + wsz)) * (2LL * H * DH);
          kreg[u] = *(const bf16x8*)(halo + hb + head * DH + su_d0[u]);
          vreg[u] = *(const bf16x8*)(halo + hb + (long long)H * DH +
                                     head * DH + su_d0[u]);
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
          float* dstv;
          float* dstk;
          if (kpos >= 0) {
            dstv = lookback
                ? dlook + look_bn + (long long)kpos * (2LL * H * DH) + lv_off
                : dacc + qkv_bn + (long long)kpos * HD3 + v_off;
            dstk = lookback
                ? dlook + look_bn + (long long)kpos * (2LL * H * DH) + lk_off
                : dacc + qkv_bn + (long long)kpos * HD3 + k_off;
          } else if (dhalo != nullptr) {
            const long long hb =
                ((long long)batch * wsz + (kpos + wsz)) * (2LL * H * DH);
            dstv = dhalo + hb + lv_off;
            dstk = dhalo + hb + lk_off;
          } else {
            continue;  // window-0 zero-pad quirk: grads discarded
          }
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

extern "C" {

void attn_bwd_launch(const void* dout, const void* qkv, const void* halo,
                     const float* rsin, const float* rcos, const void* out,
                     const float* lse, float* dacc, float* dlook,
                     float* dhalo, void* dqkv, int B, int N,
                     int H, int wsz, hipStream_t stream) {
  dim3 grid(N / wsz, H, B), block(ATTN_BLOCK);
  size_t lds = 24576 + 131072 + 2048;  // 154 KiB
  attn_bwd_kernel<<<grid, block, lds, stream>>>(
      (const short*)dout, (const short*)qkv, (const short*)halo,
      (const short*)out, lse, dacc, dlook, dhalo, B, N, H, wsz);
  long long total = (long long)B * N * 3 * H * (DH / 8);
  int fin_grid = (int)((total + 255) / 256);
  if (fin_grid > 2048) fin_grid = 2048;
  attn_bwd_finalize_kernel<<<fin_grid, 256, 0, stream>>>(
      dacc, dlook, rsin, rcos, (short*)dqkv, B, N, H, wsz);
}

}  // extern "C"
