// Fused local-window attention, forward (reference: progen.py:83-103).
//
// Input is the PRE-ROTATED qkv (ops/hip/rope_qkv.hip applies the
// reference's rotary-on-q/k/v quirk, progen.py:87, in a separate
// memory-bound pass), so this kernel's staging is a pure bf16 copy.
//
// One kernel fuses, per (batch, head, window):
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
// B-fragment ds_read_b128s are <=2-way bank conflicted. Staging uses the
// async-STAGE split (T14): tile t+1's global loads are issued before
// tile t's compute, the LDS writes land after the barrier — hiding HBM
// latency at 1 wave/SIMD occupancy. Q fragments (pre-scaled) and the
// fp32 softmax state live in registers for the whole block. Saves
// per-row logsumexp (B, h, N) fp32 for the backward's recompute.
//
// dim_head is fixed at 64 (the ProGen family's head size).

#include "common.h"

#define DH 64
#define KT 64                 // keys per tile
#define ATTN_WAVES 4
#define ATTN_BLOCK (ATTN_WAVES * WAVE)
#define NEG_INF (-1e30f)

__device__ __forceinline__ int swz(int row, int byte_in_row) {
  return (byte_in_row ^ ((row & 7) << 4));
}

__global__ __launch_bounds__(ATTN_BLOCK) void attn_fwd_kernel(
    const short* __restrict__ qkv,   // (B, N, 3*H*DH) bf16, PRE-ROTATED
    short* __restrict__ out,         // (B, N, H*DH) bf16
    float* __restrict__ lse_out,     // (B, H, N)
    int B, int N, int H, int wsz) {
  const int window = blockIdx.x;
  const int head = blockIdx.y;
  const int batch = blockIdx.z;

  const int lane = threadIdx.x % WAVE;
  const int wid = threadIdx.x / WAVE;
  const int l15 = lane & 15;
  const int l4 = lane >> 4;

  const long long HD3 = 3LL * H * DH;
  const long long qkv_bn = (long long)batch * N * HD3;
  const int q_off = head * DH;
  const int k_off = H * DH + head * DH;
  const int v_off = 2 * H * DH + head * DH;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* k_lds = smem;                          // 8 KiB
  char* v_lds = smem + KT * DH * 2;            // 8 KiB (V^T image)
  char* p_lds = smem + 2 * KT * DH * 2 + wid * 64 * KT * 2;  // 8 KiB/wave

  const float scale = rsqrtf((float)DH);
  const int tiles = 2 * wsz / KT;
  const int chunks = wsz / 64;
  const int rounds = (chunks + ATTN_WAVES - 1) / ATTN_WAVES;

  // staging geometry: 512 (key, d0) units per tile, 2 per thread
  const int su_key[2] = {(int)threadIdx.x >> 3,
                         (int)(threadIdx.x + ATTN_BLOCK) >> 3};
  const int su_d0[2] = {((int)threadIdx.x & 7) * 8,
                        (((int)threadIdx.x + ATTN_BLOCK) & 7) * 8};

  for (int round = 0; round < rounds; ++round) {
    const int chunk = round * ATTN_WAVES + wid;
    const bool active = chunk < chunks;
    const int chunk_off = chunk * 64;
    const int q0 = window * wsz + chunk_off;

    // ---- Q fragments (pre-rotated; fold in the softmax scale) ----
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
          bf16x8 o;
#pragma unroll
          for (int j = 0; j < 8; ++j)
            ((short*)&o)[j] = f2bf(bf2f(((short*)&v)[j]) * scale);
          qfrag[m][ks] = o;
        }
      }
    }

    float m_run[4][4], l_run[4][4];
#pragma unroll
    for (int m = 0; m < 4; ++m)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        m_run[m][r] = NEG_INF;
        l_run[m][r] = 0.f;
      }
    f32x4 oacc[4][4];
#pragma unroll
    for (int m = 0; m < 4; ++m)
#pragma unroll
      for (int d = 0; d < 4; ++d) oacc[m][d] = (f32x4){0.f, 0.f, 0.f, 0.f};

    const int max_tile = active ? ((chunk_off + 63 + wsz) / KT) : -1;

    // ---- T14 staging: issue loads early, write LDS after barrier ----
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
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const int d = d0 + j;
          *(short*)(v_lds + d * 128 + swz(d, key * 2)) = ((short*)&vreg[u])[j];
        }
      }
    };

    issue_loads(0);
    write_lds();
    __syncthreads();

    for (int t = 0; t < tiles; ++t) {
      if (t + 1 < tiles) issue_loads(t + 1);  // in flight during compute

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

            const float mnew = fmaxf(m_run[m][r], mx);
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
#pragma unroll
            for (int d = 0; d < 4; ++d)
              ((float*)&oacc[m][d])[r] *= alpha;
          }
        }

        // ---- P -> bf16 -> per-wave LDS ----
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

      __syncthreads();  // all waves done reading LDS tile t
      if (t + 1 < tiles) {
        write_lds();    // compiler inserts the vmcnt wait at first use
        __syncthreads();
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
    if (round + 1 < rounds) __syncthreads();
  }
}

extern "C" {

void attn_fwd_launch(const void* qkv_rot, void* out, float* lse, int B, int N,
                     int H, int wsz, hipStream_t stream) {
  dim3 grid(N / wsz, H, B), block(ATTN_BLOCK);
  size_t lds = (size_t)(2 * KT * DH * 2) + (size_t)ATTN_WAVES * 64 * KT * 2;
  attn_fwd_kernel<<<grid, block, lds, stream>>>(
      (const short*)qkv_rot, (short*)out, lse, B, N, H, wsz);
}

}  // extern "C"
