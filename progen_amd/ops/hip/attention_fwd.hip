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
//     baked into the tile iteration (fully-masked tiles skipped; fully
//     VISIBLE tiles skip the per-element compare);
//   - fp32 online softmax (max-subtract parity with progen.py:98-99);
//   - P·V accumulation and the '(w n) (h d)' output merge (progen.py:102).
//
// Geometry (CDNA4): block = 4 waves; each wave owns a 32-row Q chunk
// (MF=2 m-fragments; 193 VGPRs -> 2 waves/SIMD), sub_per_win blocks
// cover a window and round-robin chunks when wsz > 128. Per 64-key tile:
//   - QK^T is computed SWAPPED — S^T = mfma(K, Q) — so the MFMA C-layout
//     holds 4 consecutive KEYS of one q-row per register quad: the P
//     tile is written to LDS [row][key] with ds_write_b64 (the
//     non-swapped form needs 64 scattered b16 writes), and the row
//     softmax reduce is 16 in-lane values + a 2-step cross-lane shuffle;
//   - K staged in LDS [key][dh] and V transposed [dh][key], both
//     XOR-swizzled (byte ^= (row&7)<<4) -> <=2-way bank conflicts on the
//     ds_read_b128 fragments; staging is DOUBLE-BUFFERED with the
//     async-stage split (T14): tile t+1's global loads issue before
//     tile t's MFMAs and the LDS writes land in the alternate buffer
//     after them — ONE barrier per tile;
//   - per-row softmax state (m, l) lives in the 4 lanes of the row's
//     shuffle group; the O rescale factor crosses to the PV C-layout
//     rows through a tiny per-wave LDS broadcast array.
// Saves per-row logsumexp (B, h, N) fp32 for the backward's recompute.
//
// dim_head is fixed at 64 (the ProGen family's head size).

#include "common.h"

#define DH 64
#define KT 64                 // keys per tile
#define ATTN_WAVES 4
#define ATTN_BLOCK (ATTN_WAVES * WAVE)
#define MF 2                  // 16-row m-fragments per wave (32-row chunks)
#define QB (MF * 16)          // q rows per wave
#define NEG_INF (-1e30f)

__device__ __forceinline__ int swz(int row, int byte_in_row) {
  return (byte_in_row ^ ((row & 7) << 4));
}

// per-key 32-B XOR window for the LINEAR [key][dh] V image (128-B
// rows): the PV B-fragments are ds_read_b64_tr_b16 transposed reads
// whose half-wave covers key rows {kb..kb+3, kb+8..kb+11}; v(k) is a
// bijection onto 0..3 within each row parity, so the 8 rows cover all
// 64 banks (same derivation as attention_bwd.hip's dS^T image)
__device__ __forceinline__ int uk4(int k) { return ((k & 2) >> 1) | ((k & 8) >> 2); }

typedef __attribute__((__vector_size__(4 * sizeof(__bf16)))) __bf16 bf16x4t;
#define AS3 __attribute__((address_space(3)))

__global__ __launch_bounds__(ATTN_BLOCK) void attn_fwd_kernel(
    const short* __restrict__ qkv,   // (B, N, 3*H*DH) bf16, PRE-ROTATED
    const short* __restrict__ halo,  // (B, wsz, 2*H*DH) rotated [k|v]
                                     // lookback for window 0 (context
                                     // parallelism: the previous rank's
                                     // last window) or nullptr (rank 0 /
                                     // single rank: the reference's
                                     // zero-pad quirk, progen.py:90-96)
    short* __restrict__ out,         // (B, N, H*DH) bf16
    float* __restrict__ lse_out,     // (B, H, N)
    int B, int N, int H, int wsz) {
  // blocks per window: each covers ATTN_WAVES*QB q rows
  const int sub_per_win = (wsz + ATTN_WAVES * QB - 1) / (ATTN_WAVES * QB);
  const int window = blockIdx.x / sub_per_win;
  const int sub = blockIdx.x % sub_per_win;
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
  // double-buffered K and V^T tiles: stage tile t+1 into the other buffer
  // while computing tile t -> ONE barrier per tile instead of two
  char* kv_base = smem;                        // 2 x (8 + 8) KiB
  char* p_lds = smem + 4 * KT * DH * 2 + wid * QB * KT * 2;  // 4 KiB/wave
  float* bc_lds = (float*)(smem + 4 * KT * DH * 2 + ATTN_WAVES * QB * KT * 2 +
                           wid * 2 * QB * 4);  // [QB alpha | QB inv_l]/wave

  const float scale = rsqrtf((float)DH);
  const int tiles = 2 * wsz / KT;
  // chunks local to this block's 128-row slice of the window
  const int chunks_all = wsz / QB;
  const int chunks = min(ATTN_WAVES, chunks_all - sub * ATTN_WAVES);
  const int rounds = 1;

  const int su_key[2] = {(int)threadIdx.x >> 3,
                         (int)(threadIdx.x + ATTN_BLOCK) >> 3};
  const int su_d0[2] = {((int)threadIdx.x & 7) * 8,
                        (((int)threadIdx.x + ATTN_BLOCK) & 7) * 8};

  for (int round = 0; round < rounds; ++round) {
    const bool active = wid < chunks;
    const int chunk_off = (sub * ATTN_WAVES + wid) * QB;  // within window
    const int q0 = window * wsz + chunk_off;

    // ---- Q fragments (pre-rotated; fold in the softmax scale).
    // The same per-lane data serves as the mfma B operand for the
    // swapped S^T = K Q^T: lane holds Q[row l15+16n][dh 8*l4+j..]. ----
    bf16x8 qfrag[MF][2];
    if (active) {
#pragma unroll
      for (int m = 0; m < MF; ++m) {
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

    // per-lane softmax state: 4 q-rows (n*16 + l15), replicated in the
    // row's 4-lane shuffle group (l4 = 0..3)
    float m_run[MF], l_run[MF];
#pragma unroll
    for (int n = 0; n < MF; ++n) {
      m_run[n] = NEG_INF;
      l_run[n] = 0.f;
    }
    f32x4 oacc[MF][4];  // [m rowblock][dh frag], C rows = l4*4+r
#pragma unroll
    for (int m = 0; m < MF; ++m)
#pragma unroll
      for (int d = 0; d < 4; ++d) oacc[m][d] = (f32x4){0.f, 0.f, 0.f, 0.f};

    const int max_tile = active ? ((chunk_off + QB - 1 + wsz) / KT) : -1;

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
    auto write_lds = [&](int buf) {
      char* k_lds = kv_base + buf * (2 * KT * DH * 2);
      char* v_lds = k_lds + KT * DH * 2;
#pragma unroll
      for (int u = 0; u < 2; ++u) {
        const int key = su_key[u];
        const int d0 = su_d0[u];
        *(bf16x8*)(k_lds + key * 128 + swz(key, d0 * 2)) = kreg[u];
        // V stored LINEAR [key][dh] (one 16-B write; the old [dh][key]
        // scatter-transpose was 8 b16 writes per fragment) — the PV
        // step reads it transposed with ds_read_b64_tr_b16 (T10)
        *(bf16x8*)(v_lds + key * 128 + ((d0 * 2) ^ (uk4(key) * 32))) = vreg[u];
      }
    };

    issue_loads(0);
    write_lds(0);
    __syncthreads();

    for (int t = 0; t < tiles; ++t) {
      char* k_lds = kv_base + (t & 1) * (2 * KT * DH * 2);
      char* v_lds = k_lds + KT * DH * 2;
      if (t + 1 < tiles) issue_loads(t + 1);

      if (active && t <= max_tile) {
        const int kb = t * KT;
        // every key of the tile visible to every row of the chunk?
        const bool tile_full = (kb + KT - 1) <= chunk_off + wsz;

        // ---- S^T = K Q^T: st[km][n] rows=keys, cols=q-rows ----
        f32x4 st[4][MF];
#pragma unroll
        for (int km = 0; km < 4; ++km)
#pragma unroll
          for (int n = 0; n < MF; ++n) st[km][n] = (f32x4){0.f, 0.f, 0.f, 0.f};
        __builtin_amdgcn_s_setprio(1);  // favor the MFMA cluster (T5)
#pragma unroll
        for (int ks = 0; ks < 2; ++ks) {
#pragma unroll
          for (int km = 0; km < 4; ++km) {
            const int key = km * 16 + l15;
            const int d0 = ks * 32 + 8 * l4;
            bf16x8 kfrag = *(const bf16x8*)(k_lds + key * 128 + swz(key, d0 * 2));
#pragma unroll
            for (int n = 0; n < MF; ++n)
              st[km][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                  kfrag, qfrag[n][ks], st[km][n], 0, 0, 0);
          }
        }
        __builtin_amdgcn_s_setprio(0);

        // ---- mask + per-row max (in-lane over 16 keys, then x-lane) ----
        float tile_max[MF];
#pragma unroll
        for (int n = 0; n < MF; ++n) {
          const int rowiw = chunk_off + n * 16 + l15;
          float mx = NEG_INF;
#pragma unroll
          for (int km = 0; km < 4; ++km)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
              float v = ((float*)&st[km][n])[r];
              if (!tile_full) {
                const int kpos_band = kb + km * 16 + l4 * 4 + r;
                if (kpos_band > rowiw + wsz) v = NEG_INF;  // progen.py:95
                ((float*)&st[km][n])[r] = v;
              }
              mx = fmaxf(mx, v);
            }
          // row data lives in lanes l15, l15+16, l15+32, l15+48
          mx = fmaxf(mx, __shfl_xor(mx, 16, 64));
          mx = fmaxf(mx, __shfl_xor(mx, 32, 64));
          tile_max[n] = mx;
        }

        // ---- online softmax update; write P (bf16) with b64 ----
#pragma unroll
        for (int n = 0; n < MF; ++n) {
          const float mnew = fmaxf(m_run[n], tile_max[n]);
          const float alpha =
              (m_run[n] == NEG_INF) ? 0.f : __expf(m_run[n] - mnew);
          float psum = 0.f;
          const int row = n * 16 + l15;
#pragma unroll
          for (int km = 0; km < 4; ++km) {
            short pk[4];
#pragma unroll
            for (int r = 0; r < 4; ++r) {
              float v = ((float*)&st[km][n])[r];
              float p = (v <= NEG_INF) ? 0.f : __expf(v - mnew);
              psum += p;
              pk[r] = f2bf(p);
            }
            *(unsigned long long*)(p_lds + row * 128 +
                                   swz(row, (km * 16 + l4 * 4) * 2)) =
                *(unsigned long long*)pk;
          }
          psum += __shfl_xor(psum, 16, 64);
          psum += __shfl_xor(psum, 32, 64);
          l_run[n] = l_run[n] * alpha + psum;
          m_run[n] = mnew;
          if (l4 == 0) bc_lds[row] = alpha;  // broadcast to PV C-layout rows
        }
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");

        // ---- O = O * alpha + P V ----
#pragma unroll
        for (int m = 0; m < MF; ++m) {
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const float alpha = bc_lds[m * 16 + l4 * 4 + r];
#pragma unroll
            for (int d = 0; d < 4; ++d)
              ((float*)&oacc[m][d])[r] *= alpha;
          }
        }
        __builtin_amdgcn_s_setprio(1);  // favor the MFMA cluster (T5)
#pragma unroll
        for (int ks = 0; ks < 2; ++ks) {
          // V B-fragments by hardware transpose read: key rows
          // kk0 + (l15>>2) (+4), column = the dh block
          const int kr1 = ks * 32 + l4 * 8 + (l15 >> 2);
          const int kr2 = kr1 + 4;
          bf16x8 vfr[4];
#pragma unroll
          for (int d = 0; d < 4; ++d) {
            const int colb = (d * 16 + (l15 & 3) * 4) * 2;
            auto p1 = (AS3 bf16x4t*)(v_lds + kr1 * 128 +
                                     (colb ^ (uk4(kr1) * 32)));
            auto p2 = (AS3 bf16x4t*)(v_lds + kr2 * 128 +
                                     (colb ^ (uk4(kr2) * 32)));
            bf16x4t a = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p1);
            bf16x4t b = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p2);
#pragma unroll
            for (int j = 0; j < 4; ++j) {
              ((__bf16*)&vfr[d])[j] = a[j];
              ((__bf16*)&vfr[d])[j + 4] = b[j];
            }
          }
#pragma unroll
          for (int m = 0; m < MF; ++m) {
            const int row = m * 16 + l15;
            const int kk0 = ks * 32 + 8 * l4;
            bf16x8 pfrag = *(const bf16x8*)(p_lds + row * 128 + swz(row, kk0 * 2));
#pragma unroll
            for (int d = 0; d < 4; ++d) {
              oacc[m][d] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                  pfrag, vfr[d], oacc[m][d], 0, 0, 0);
            }
          }
        }
        __builtin_amdgcn_s_setprio(0);
      }

      // stage tile t+1 into the OTHER buffer: no reader conflict (its
      // last readers finished before the barrier that ended tile t-1)
      if (t + 1 < tiles) write_lds((t + 1) & 1);
      __syncthreads();
    }

    // ---- epilogue: O /= l, store out + lse ----
    if (active) {
      // broadcast inv_l and write lse from the softmax-state lanes
#pragma unroll
      for (int n = 0; n < MF; ++n) {
        const int row = n * 16 + l15;
        if (l4 == 0) {
          bc_lds[QB + row] = 1.0f / l_run[n];
          lse_out[((long long)batch * H + head) * N + q0 + row] =
              m_run[n] + logf(l_run[n]);
        }
      }
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      const long long out_bn = ((long long)batch * N) * (long long)(H * DH);
#pragma unroll
      for (int m = 0; m < MF; ++m) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int row = q0 + m * 16 + l4 * 4 + r;
          const float inv_l = bc_lds[QB + m * 16 + l4 * 4 + r];
#pragma unroll
          for (int d = 0; d < 4; ++d) {
            const int dcol = d * 16 + l15;
            out[out_bn + (long long)row * (H * DH) + head * DH + dcol] =
                f2bf(((float*)&oacc[m][d])[r] * inv_l);
          }
        }
      }
    }
    if (rounds > 1) __syncthreads();
  }
}

extern "C" {

void attn_fwd_launch(const void* qkv_rot, const void* halo, void* out,
                     float* lse, int B, int N, int H, int wsz,
                     hipStream_t stream) {
  const int sub_per_win = (wsz + ATTN_WAVES * QB - 1) / (ATTN_WAVES * QB);
  dim3 grid((N / wsz) * sub_per_win, H, B), block(ATTN_BLOCK);
  size_t lds = (size_t)(4 * KT * DH * 2) + (size_t)ATTN_WAVES * QB * KT * 2 +
               ATTN_WAVES * 2 * QB * 4;
  attn_fwd_kernel<<<grid, block, lds, stream>>>(
      (const short*)qkv_rot, (const short*)halo, (short*)out, lse, B, N, H,
      wsz);
}

}  // extern "C"
