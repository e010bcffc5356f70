// Fused local-window attention, forward (reference: progen.py:83-103).
//
// One kernel fuses, per (batch, head, window):
//   - GPT-J interleaved rotary applied to q, k AND v (quirk preserved,
//     reference: progen.py:87) during register/LDS staging;
//   - the one-window-lookback key band [prev window ‖ own window] with
//     window 0's lookback keys ZERO and UNMASKED (progen.py:90-96):
//     zero K rows give logit 0 into the softmax denominator and zero V
//     rows contribute nothing — exact parity by zero-filling the tiles;
//   - the offset-causal mask tril(ones(wsz, 2wsz), k=wsz) (progen.py:95)
//     baked into the tile iteration (fully-masked tiles skipped);
//   - fp32 online softmax (max-subtract parity with progen.py:98-99);
//   - P·V accumulation and the '(w n) (h d)' output merge (progen.py:102).
//
// Geometry (CDNA4): block = 4 waves = one window; each wave owns a
// 64-row Q chunk (round-robins chunks when wsz > 256). Per 64-key tile:
// K staged in LDS [key][dh] and V transposed [dh][key], both
// XOR-swizzled (byte ^= (row&7)<<4) so the mfma_f32_16x16x32_bf16
// B-fragment ds_read_b128s are <=2-way bank conflicted. Q fragments and
// the fp32 softmax state live in registers for the whole block.
// Saves per-row logsumexp (B, h, N) fp32 for the backward's recompute.
//
// dim_head is fixed at 64 (the ProGen family's head size).

#include "common.h"

#define DH 64
#define KT 64                 // keys per tile
#define ATTN_WAVES 4
#define ATTN_BLOCK (ATTN_WAVES * WAVE)
#define NEG_INF (-1e30f)

using f32x4v = f32x4;

__device__ __forceinline__ int swz(int row, int byte_in_row) {
  return (byte_in_row ^ ((row & 7) << 4));
}

// rotary: interleaved pairs (progen.py:30-41); sin/cos tables are
// repeat-interleaved so sin[2i] == sin[2i+1]
__device__ __forceinline__ void rope8(float* x, const float* sinv,
                                      const float* cosv) {
#pragma unroll
  for (int p = 0; p < 4; ++p) {
    float x0 = x[2 * p], x1 = x[2 * p + 1];
    float s = sinv[2 * p], c = cosv[2 * p];
    x[2 * p] = x0 * c - x1 * s;
    x[2 * p + 1] = x1 * c + x0 * s;
  }
}

__global__ __launch_bounds__(ATTN_BLOCK) void attn_fwd_kernel(
    const short* __restrict__ qkv,   // (B, N, 3*H*DH) bf16
    const float* __restrict__ rsin,  // (N, DH)
    const float* __restrict__ rcos,  // (N, DH)
    short* __restrict__ out,         // (B, N, H*DH) bf16
    float* __restrict__ lse_out,     // (B, H, N)
    int B, int N, int H, int wsz) {
  const int window = blockIdx.x;
  const int head = blockIdx.y;
  const int batch = blockIdx.z;
  const int nwin = N / wsz;
  (void)nwin;

  const int lane = threadIdx.x % WAVE;
  const int wid = threadIdx.x / WAVE;
  const int l15 = lane & 15;
  const int l4 = lane >> 4;

  const long long HD3 = 3LL * H * DH;
  const long long qkv_bn = (long long)batch * N * HD3;
  const int q_off = head * DH;
  const int k_off = H * DH + head * DH;
  const int v_off = 2 * H * DH + head * DH;

  // LDS: K tile [KT][DH] swizzled + V^T tile [DH][KT] swizzled + per-wave
  // P [64][KT] swizzled
  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* k_lds = smem;                          // KT*DH*2 = 8 KiB
  char* v_lds = smem + KT * DH * 2;            // 8 KiB
  char* p_lds = smem + 2 * KT * DH * 2 + wid * 64 * KT * 2;  // 8 KiB/wave

  const float scale = rsqrtf((float)DH);
  const int tiles = 2 * wsz / KT;
  const int chunks = wsz / 64;  // 64-row q chunks in this window
  const int rounds = (chunks + ATTN_WAVES - 1) / ATTN_WAVES;

  for (int round = 0; round < rounds; ++round) {
    const int chunk = round * ATTN_WAVES + wid;
    const bool active = chunk < chunks;
    const int chunk_off = chunk * 64;  // q-row offset within the window
    const int q0 = window * wsz + chunk_off;  // global q row of this wave

    // ---- load Q fragments (+rotary, *scale) into registers ----
    bf16x8 qfrag[4][2];
    if (active) {
#pragma unroll
      for (int m = 0; m < 4; ++m) {
        const int row = q0 + m * 16 + l15;
#pragma unroll
        for (int ks = 0; ks < 2; ++ks) {
          const int d0 = ks * 32 + 8 * l4;
          bf16x8 v = *(const bf16x8*)(qkv + qkv_bn + (long long)row * HD3 +
                                      q_off + d0);
          float x[8], sv[8], cv[8];
#pragma unroll
          for (int j = 0; j < 8; ++j) x[j] = bf2f(((short*)&v)[j]);
          *(f32x4*)(sv) = *(const f32x4*)(rsin + (long long)row * DH + d0);
          *(f32x4*)(sv + 4) = *(const f32x4*)(rsin + (long long)row * DH + d0 + 4);
          *(f32x4*)(cv) = *(const f32x4*)(rcos + (long long)row * DH + d0);
          *(f32x4*)(cv + 4) = *(const f32x4*)(rcos + (long long)row * DH + d0 + 4);
          rope8(x, sv, cv);
          bf16x8 o;
#pragma unroll
          for (int j = 0; j < 8; ++j) ((short*)&o)[j] = f2bf(x[j] * scale);
          qfrag[m][ks] = o;
        }
      }
    }

    // ---- softmax state ----
    float m_run[4][4], l_run[4][4];
#pragma unroll
    for (int m = 0; m < 4; ++m)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        m_run[m][r] = NEG_INF;
        l_run[m][r] = 0.f;
      }
    f32x4 oacc[4][4];  // [m][dh fragment] per-lane 4 rows x 1 col
#pragma unroll
    for (int m = 0; m < 4; ++m)
#pragma unroll
      for (int d = 0; d < 4; ++d) oacc[m][d] = (f32x4){0.f, 0.f, 0.f, 0.f};

    const int max_tile = active ? ((chunk_off + 63 + wsz) / KT) : -1;

    for (int t = 0; t < tiles; ++t) {
      // ---- cooperative stage of K tile and V^T tile (+rotary) ----
      // 256 threads x 16 B = 4 KiB per pass; tile is 8 KiB -> 2 passes
      __syncthreads();
#pragma unroll
      for (int pass = 0; pass < 2; ++pass) {
        const int flat = pass * ATTN_BLOCK + threadIdx.x;  // 0..511
        const int key = flat >> 3;           // 0..63 within tile
        const int d0 = (flat & 7) * 8;       // dh group of 8
        const int kpos_band = t * KT + key;  // 0..2wsz
        const int kpos = (window - 1) * wsz + kpos_band;  // global key pos
        float kx[8], vx[8];
        if (kpos >= 0) {
          bf16x8 kvec = *(const bf16x8*)(qkv + qkv_bn + (long long)kpos * HD3 +
                                         k_off + d0);
          bf16x8 vvec = *(const bf16x8*)(qkv + qkv_bn + (long long)kpos * HD3 +
                                         v_off + d0);
          float sv[8], cv[8];
          *(f32x4*)(sv) = *(const f32x4*)(rsin + (long long)kpos * DH + d0);
          *(f32x4*)(sv + 4) = *(const f32x4*)(rsin + (long long)kpos * DH + d0 + 4);
          *(f32x4*)(cv) = *(const f32x4*)(rcos + (long long)kpos * DH + d0);
          *(f32x4*)(cv + 4) = *(const f32x4*)(rcos + (long long)kpos * DH + d0 + 4);
#pragma unroll
          for (int j = 0; j < 8; ++j) kx[j] = bf2f(((short*)&kvec)[j]);
#pragma unroll
          for (int j = 0; j < 8; ++j) vx[j] = bf2f(((short*)&vvec)[j]);
          rope8(kx, sv, cv);
          rope8(vx, sv, cv);  // rotary on V too (progen.py:87)
        } else {
          // window 0 lookback: the zero-pad window (progen.py:90-91)
#pragma unroll
          for (int j = 0; j < 8; ++j) kx[j] = vx[j] = 0.f;
        }
        bf16x8 kb;
#pragma unroll
        for (int j = 0; j < 8; ++j) ((short*)&kb)[j] = f2bf(kx[j]);
        *(bf16x8*)(k_lds + key * 128 + swz(key, d0 * 2)) = kb;
        // V^T: scatter 8 bf16 to [d][key]
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const int d = d0 + j;
          *(short*)(v_lds + d * 128 + swz(d, key * 2)) = f2bf(vx[j]);
        }
      }
      __syncthreads();

      if (active && t <= max_tile) {
        // ---- S = Q K^T ----
        f32x4 s[4][4];
#pragma unroll
        for (int m = 0; m < 4; ++m)
#pragma unroll
          for (int n = 0; n < 4; ++n) s[m][n] = (f32x4){0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int ks = 0; ks < 2; ++ks) {
#pragma unroll
          for (int n = 0; n < 4; ++n) {
            const int key = n * 16 + l15;
            const int d0 = ks * 32 + 8 * l4;
            bf16x8 kfrag = *(const bf16x8*)(k_lds + key * 128 + swz(key, d0 * 2));
#pragma unroll
            for (int m = 0; m < 4; ++m)
              s[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                  qfrag[m][ks], kfrag, s[m][n], 0, 0, 0);
          }
        }

        // ---- mask + online softmax ----
        const int kb = t * KT;
        float tile_max[4][4];
#pragma unroll
        for (int m = 0; m < 4; ++m) {
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const int rowiw = chunk_off + m * 16 + l4 * 4 + r;
            float mx = NEG_INF;
#pragma unroll
            for (int n = 0; n < 4; ++n) {
              const int kpos_band = kb + n * 16 + l15;
              float v = ((float*)&s[m][n])[r];
              if (kpos_band > rowiw + wsz) v = NEG_INF;  // progen.py:95
              ((float*)&s[m][n])[r] = v;
              mx = fmaxf(mx, v);
            }
            mx = group16_max(mx);
            tile_max[m][r] = mx;
          }
        }
#pragma unroll
        for (int m = 0; m < 4; ++m) {
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const float mnew = fmaxf(m_run[m][r], tile_max[m][r]);
            const float alpha =
                (m_run[m][r] == NEG_INF) ? 0.f : __expf(m_run[m][r] - mnew);
            float psum = 0.f;
#pragma unroll
            for (int n = 0; n < 4; ++n) {
              float v = ((float*)&s[m][n])[r];
              float p = (v == NEG_INF) ? 0.f : __expf(v - mnew);
              ((float*)&s[m][n])[r] = p;
              psum += p;
            }
            psum = group16_sum(psum);
            l_run[m][r] = l_run[m][r] * alpha + psum;
            m_run[m][r] = mnew;
            // rescale O rows
#pragma unroll
            for (int d = 0; d < 4; ++d)
              ((float*)&oacc[m][d])[r] *= alpha;
          }
        }

        // ---- P -> bf16 -> LDS (per-wave region) ----
#pragma unroll
        for (int m = 0; m < 4; ++m)
#pragma unroll
          for (int n = 0; n < 4; ++n) {
            const int key = n * 16 + l15;
#pragma unroll
            for (int r = 0; r < 4; ++r) {
              const int row = m * 16 + l4 * 4 + r;
              *(short*)(p_lds + row * 128 + swz(row, key * 2)) =
                  f2bf(((float*)&s[m][n])[r]);
            }
          }
        // wave-local LDS write->read (no cross-wave sharing of p_lds):
        // drain DS writes before the fragment reads; a block barrier is
        // illegal here (divergent path), a wave-local wait suffices.
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");

        // ---- O += P V ----
#pragma unroll
        for (int ks = 0; ks < 2; ++ks) {
#pragma unroll
          for (int m = 0; m < 4; ++m) {
            const int row = m * 16 + l15;
            const int kk0 = ks * 32 + 8 * l4;
            bf16x8 pfrag = *(const bf16x8*)(p_lds + row * 128 + swz(row, kk0 * 2));
#pragma unroll
            for (int d = 0; d < 4; ++d) {
              const int dcol = d * 16 + l15;
              bf16x8 vfrag = *(const bf16x8*)(v_lds + dcol * 128 + swz(dcol, kk0 * 2));
              oacc[m][d] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                  pfrag, vfrag, oacc[m][d], 0, 0, 0);
            }
          }
        }
      }
    }

    // ---- epilogue: O /= l, store out + lse ----
    if (active) {
      const long long out_bn = ((long long)batch * N) * (long long)(H * DH);
#pragma unroll
      for (int m = 0; m < 4; ++m) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int row = q0 + m * 16 + l4 * 4 + r;
          const float inv_l = 1.0f / l_run[m][r];
#pragma unroll
          for (int d = 0; d < 4; ++d) {
            const int dcol = d * 16 + l15;
            out[out_bn + (long long)row * (H * DH) + head * DH + dcol] =
                f2bf(((float*)&oacc[m][d])[r] * inv_l);
          }
          if (l15 == 0) {
            lse_out[((long long)batch * H + head) * N + row] =
                m_run[m][r] + logf(l_run[m][r]);
          }
        }
      }
    }
  }
}

extern "C" {

void attn_fwd_launch(const void* qkv, const float* rsin, const float* rcos,
                     void* out, float* lse, int B, int N, int H, int wsz,
                     hipStream_t stream) {
  dim3 grid(N / wsz, H, B), block(ATTN_BLOCK);
  size_t lds = (size_t)(2 * KT * DH * 2) + (size_t)ATTN_WAVES * 64 * KT * 2;
  attn_fwd_kernel<<<grid, block, lds, stream>>>(
      (const short*)qkv, rsin, rcos, (short*)out, lse, B, N, H, wsz);
}

}  // extern "C"
