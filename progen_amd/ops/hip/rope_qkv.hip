// Pre-rotation of the QKV projection: applies GPT-J interleaved rotary
// to q, k AND v (the reference quirk, progen.py:87) in one memory-bound
// pass, so the attention kernels stage pure bf16 copies with no
// dependent sin/cos loads or rope VALU work on their critical path
// (at 1 wave/SIMD those latency chains dominated the fused kernels).
//
// The inverse rotation lives in attn_bwd_finalize_kernel
// (attention_bwd.hip) — rotary is linear, so d(qkv) = R^-1(d(qkv_rot)).

#include "common.h"

#define DH 64

__global__ __launch_bounds__(256) void rope_qkv_kernel(
    const short* __restrict__ qkv,  // (B, N, 3*H*DH) bf16
    const float* __restrict__ rsin, const float* __restrict__ rcos,
    short* __restrict__ qkv_rot, int B, int N, int H) {
  const long long HD3 = 3LL * H * DH;
  // grid.x covers one (B,N)-row's 3H*8 vector units CONTIGUOUSLY
  // (hslot/g by shift+mask — the old flat form paid two 64-bit
  // divisions by the non-power-of-two 3H per iteration; a first 2-D
  // rework put g in the low lane bits and measured 23% SLOWER from the
  // scattered access, so the in-row order stays sequential), grid.y
  // stripes the rows
  const int u = blockIdx.x * 256 + (int)threadIdx.x;
  if (u >= 3 * H * (DH / 8)) return;
  const int hslot = u >> 3;
  const int g = u & 7;
  const int d0 = g * 8;
  const long long BN = (long long)B * N;
  for (long long bn = blockIdx.y; bn < BN; bn += gridDim.y) {
    const int n = (int)(bn % N);

    const long long off = bn * HD3 + (long long)hslot * DH + d0;
    bf16x8 v = *(const bf16x8*)(qkv + off);
    float x[8], sv[8], cv[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) x[j] = bf2f(((short*)&v)[j]);
    *(f32x4*)(sv) = *(const f32x4*)(rsin + (long long)n * DH + d0);
    *(f32x4*)(sv + 4) = *(const f32x4*)(rsin + (long long)n * DH + d0 + 4);
    *(f32x4*)(cv) = *(const f32x4*)(rcos + (long long)n * DH + d0);
    *(f32x4*)(cv + 4) = *(const f32x4*)(rcos + (long long)n * DH + d0 + 4);
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      float x0 = x[2 * p], x1 = x[2 * p + 1];
      float s = sv[2 * p], c = cv[2 * p];
      x[2 * p] = x0 * c - x1 * s;
      x[2 * p + 1] = x1 * c + x0 * s;
    }
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) ((short*)&o)[j] = f2bf(x[j]);
    *(bf16x8*)(qkv_rot + off) = o;
  }
}

extern "C" {

void rope_qkv_launch(const void* qkv, const float* rsin, const float* rcos,
                     void* qkv_rot, int B, int N, int H, hipStream_t stream) {
  int gx = (3 * H * (DH / 8) + 255) / 256;
  long long bn = (long long)B * N;
  int gy = (int)(bn < 2048 ? bn : 2048);
  dim3 grid(gx, gy);
  rope_qkv_kernel<<<grid, 256, 0, stream>>>((const short*)qkv, rsin, rcos,
                                            (short*)qkv_rot, B, N, H);
}

}  // extern "C"
