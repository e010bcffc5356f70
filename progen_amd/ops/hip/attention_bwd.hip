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
// Geometry: block = one (batch, head, window), EIGHT waves of 32-row
// q-chunks (round-2 ladder, tools/ablate_attn_bwd.hip: the V4 2-barrier
// 4-wave restructure took 1965 -> 1500 us/call at the production grid;
// this V5 8-wave geometry takes it to 1257 us, both bitwise-identical.
// 8 waves halve the per-wave register state (s/dp/dqacc are [2][4]:
// 240 VGPRs) so each SIMD runs TWO waves that cover each other's
// stalls — the V4 PMC still showed 47.8% of cycles barrier/wait-parked
// at 1 wave/SIMD). Wave pairs share the 64-row chunk LDS regions
// (layouts and swizzles unchanged); the dV/dK key slices split their
// MFMA K-dim by ks-half across the block's wave halves, the upper half
// parking its partials in the (dead-by-then) dS^T region and the lower
// half combining + storing (plain stores stay raceless, no extra
// global buffers). Per 64-key tile —
//   phase 1 (wave-local, no barrier): S, P (P^T b64-written), dP,
//     dS = P o (dP - D) (dS^T b64-written), and dQ += dS k_s where the
//     dS A-fragments come from ds_read_b64_tr_b16 transposed reads of
//     the wave's OWN just-written dS^T region (same-wave DS ordering is
//     program order).
//   barrier A; tile t+1's k/v/kt staging overlaps the dV/dK slices
//   (each wave owns a 16-key x 32-dh output slice — the 8 waves split
//   a slice by DH-half, summing the full K-dim, so outputs stay
//   disjoint with no cross-wave combine); barrier B. Two barriers per
//   tile (an earlier ks-half split needed an LDS partial combine and
//   four barriers; PMC showed 53% of cycles parked on them).
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

#define V5_WAVES 8
#define V5_BLOCK (V5_WAVES * WAVE)

__global__ __launch_bounds__(V5_BLOCK) void attn_bwd_kernel(
    const short* __restrict__ dout, const short* __restrict__ qkv,
    const short* __restrict__ halo,   // CP lookback for window 0 (see
                                      // attention_fwd.hip) or nullptr
    const short* __restrict__ out, const float* __restrict__ lse,
    float* __restrict__ dacc, float* __restrict__ dlook,
    float* __restrict__ dhalo,        // (B, wsz, 2*H*DH) fp32 window-0
                                      // lookback grads (CP) or nullptr
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
      } else if (halo != nullptr) {
        const long long hb =
            ((long long)batch * wsz + (kpos + wsz)) * (2LL * H * DH);
        kreg = *(const bf16x8*)(halo + hb + head * DH + su_d0);
        vreg = *(const bf16x8*)(halo + hb + (long long)H * DH + head * DH +
                                su_d0);
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

      // ---- phase 2: dV+dK key slices. The 8 waves split each 16-key
      // slice by DH-HALF (32 channels each), every wave summing the
      // FULL K-dim — outputs are disjoint, so the stores stay plain
      // and no cross-wave partial combine (or its two extra barriers)
      // is needed (the earlier ks-half split parked 53% of cycles on
      // its 4-barriers-per-tile schedule). ----
      {
        const int dh0 = (wid >> 2) * 2;   // wave's 2 d-blocks (32 dh)
        const int keyslot = (wid & 3) * 16 + l15;
        f32x4 dv[2], dk[2];
#pragma unroll
        for (int n = 0; n < 2; ++n) {
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
            const int r0 = ks * 32 + 8 * l4;
            bf16x8 pf = *(const bf16x8*)(pds_c + keyslot * 128 +
                                         swz(keyslot, r0 * 2));
            bf16x8 dsf = *(const bf16x8*)(ds2_c + keyslot * 128 +
                                          ((r0 * 2) ^ (uk4(keyslot) * 32)));
#pragma unroll
            for (int n = 0; n < 2; ++n) {
              const int d = (dh0 + n) * 16 + l15;
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
          const int kpos = (window - 1) * wsz + kb + (wid & 3) * 16 + l4 * 4 + r;
          float* dstv;
          float* dstk;
          if (kpos >= 0) {
            dstv = lookback
                ? dlook + look_bn + (long long)kpos * HD2 + lv_off
                : dacc + qkv_bn + (long long)kpos * HD3 + v_off;
            dstk = lookback
                ? dlook + look_bn + (long long)kpos * HD2 + lk_off
                : dacc + qkv_bn + (long long)kpos * HD3 + k_off;
          } else if (dhalo != nullptr) {
            const long long hb =
                ((long long)batch * wsz + (kpos + wsz)) * HD2;
            dstv = dhalo + hb + lv_off;
            dstk = dhalo + hb + lk_off;
          } else {
            continue;  // window-0 zero-pad quirk: grads discarded
          }
#pragma unroll
          for (int n = 0; n < 2; ++n) {
            const int dcol = (dh0 + n) * 16 + l15;
            float vv = ((float*)&dv[n])[r];
            float vk = ((float*)&dk[n])[r];
            if (round > 0) {
              vv += dstv[dcol];
              vk += dstk[dcol];
            }
            dstv[dcol] = vv;
            dstk[dcol] = vk;
          }
        }
      }
      __syncthreads();  // barrier B: slices done before t+1's phase 1
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
  // grid.x covers one row's 3H*8 vector units contiguously (hslot/g by
  // shift+mask: no 64-bit division, no scattered access — see
  // rope_qkv.hip), grid.y stripes the (B,N) rows
  const int u = blockIdx.x * 256 + (int)threadIdx.x;
  if (u >= 3 * H * (DH / 8)) return;
  const int hslot = u >> 3;
  const int g = u & 7;
  const int d0 = g * 8;
  const long long BN = (long long)B * N;
  for (long long bn = blockIdx.y; bn < BN; bn += gridDim.y) {
    const int n = (int)(bn % N);

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
  dim3 grid(N / wsz, H, B), block(V5_BLOCK);  // 8 waves: 2/SIMD
  size_t lds = 24576 + 131072 + 2048;  // 154 KiB
  attn_bwd_kernel<<<grid, block, lds, stream>>>(
      (const short*)dout, (const short*)qkv, (const short*)halo,
      (const short*)out, lse, dacc, dlook, dhalo, B, N, H, wsz);
  int gx = (3 * H * (DH / 8) + 255) / 256;
  long long bn_tot = (long long)B * N;
  int gy = (int)(bn_tot < 2048 ? bn_tot : 2048);
  dim3 fin_grid(gx, gy);
  attn_bwd_finalize_kernel<<<fin_grid, 256, 0, stream>>>(
      dacc, dlook, rsin, rcos, (short*)dqkv, B, N, H, wsz);
}

}  // extern "C"
