// Fused log-softmax + NLL gather for the masked cross-entropy
// (reference: utils.py:45-59). The (B, N)-shaped EOS/pad mask reduction
// stays on the torch side (ops/functional.py); this kernel owns the
// (B, N, V) work in one pass per row:
//   fwd: nll[row] = lse(logits[row]) - logits[row][target]
//   bwd: dlogits[row][v] = dnll[row] * (softmax(v) - onehot[target])
// One wave per row (V = num_tokens = 256 -> 4 elements per lane).

#include "common.h"

#define CE_WAVES 4
#define CE_BLOCK (CE_WAVES * WAVE)

template <bool IS_BF16>
__global__ __launch_bounds__(CE_BLOCK) void ce_fwd_kernel(
    const void* __restrict__ logits_, const long long* __restrict__ targets,
    float* __restrict__ nll, float* __restrict__ lse_out, long long R, int V) {
  const long long row = blockIdx.x * (long long)CE_WAVES + threadIdx.x / WAVE;
  if (row >= R) return;
  const int lane = threadIdx.x % WAVE;

  const short* lb = IS_BF16 ? ((const short*)logits_) + row * V : nullptr;
  const float* lf = IS_BF16 ? nullptr : ((const float*)logits_) + row * V;

  float m = -INFINITY;
  for (int v = lane; v < V; v += WAVE) {
    float x = IS_BF16 ? bf2f(lb[v]) : lf[v];
    m = fmaxf(m, x);
  }
  m = wave_max(m);

  float s = 0.f;
  for (int v = lane; v < V; v += WAVE) {
    float x = IS_BF16 ? bf2f(lb[v]) : lf[v];
    s += __expf(x - m);
  }
  s = wave_sum(s);
  const float lse = logf(s) + m;

  if (lane == 0) {
    const long long t = targets[row];
    const float xt = IS_BF16 ? bf2f(lb[t]) : lf[t];
    nll[row] = lse - xt;
    lse_out[row] = lse;
  }
}

template <bool IS_BF16>
__global__ __launch_bounds__(CE_BLOCK) void ce_bwd_kernel(
    const float* __restrict__ dnll, const void* __restrict__ logits_,
    const long long* __restrict__ targets, const float* __restrict__ lse,
    void* __restrict__ dlogits_, long long R, int V) {
  const long long row = blockIdx.x * (long long)CE_WAVES + threadIdx.x / WAVE;
  if (row >= R) return;
  const int lane = threadIdx.x % WAVE;

  const short* lb = IS_BF16 ? ((const short*)logits_) + row * V : nullptr;
  const float* lf = IS_BF16 ? nullptr : ((const float*)logits_) + row * V;
  short* db = IS_BF16 ? ((short*)dlogits_) + row * V : nullptr;
  float* df = IS_BF16 ? nullptr : ((float*)dlogits_) + row * V;

  const float d = dnll[row];
  const float l = lse[row];
  const long long t = targets[row];

  for (int v = lane; v < V; v += WAVE) {
    float x = IS_BF16 ? bf2f(lb[v]) : lf[v];
    float g = d * (__expf(x - l) - (v == (int)t ? 1.f : 0.f));
    if (IS_BF16) db[v] = f2bf(g); else df[v] = g;
  }
}

extern "C" {

void ce_fwd_launch(const void* logits, const long long* targets, float* nll,
                   float* lse, long long R, int V, bool is_bf16,
                   hipStream_t stream) {
  long long blocks = (R + CE_WAVES - 1) / CE_WAVES;
  if (is_bf16)
    ce_fwd_kernel<true><<<blocks, CE_BLOCK, 0, stream>>>(logits, targets, nll,
                                                         lse, R, V);
  else
    ce_fwd_kernel<false><<<blocks, CE_BLOCK, 0, stream>>>(logits, targets, nll,
                                                          lse, R, V);
}

void ce_bwd_launch(const float* dnll, const void* logits,
                   const long long* targets, const float* lse, void* dlogits,
                   long long R, int V, bool is_bf16, hipStream_t stream) {
  long long blocks = (R + CE_WAVES - 1) / CE_WAVES;
  if (is_bf16)
    ce_bwd_kernel<true><<<blocks, CE_BLOCK, 0, stream>>>(dnll, logits, targets,
                                                         lse, dlogits, R, V);
  else
    ce_bwd_kernel<false><<<blocks, CE_BLOCK, 0, stream>>>(dnll, logits, targets,
                                                          lse, dlogits, R, V);
}

}  // extern "C"
