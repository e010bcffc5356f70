// Fused global-norm-clip + AdamW over the flat parameter space
// (reference semantics: optax chain at train.py:115-121).
//
// One kernel pass updates fp32 master weights, fp32 moments and the
// bf16 model params from the (accumulated) flat gradient buffer. The
// per-chunk tables carry the ndim>1 weight-decay mask (train.py:115);
// the clip coefficient is a device scalar produced by the grad-norm
// reduction so no host sync is needed.

#include "common.h"

#define ADAMW_BLOCK 256

template <bool IS_BF16>
__global__ __launch_bounds__(ADAMW_BLOCK) void fused_adamw_kernel(
    float* __restrict__ master, void* __restrict__ params,
    const void* __restrict__ grads, float* __restrict__ exp_avg,
    float* __restrict__ exp_avg_sq, const long long* __restrict__ starts,
    const long long* __restrict__ ends, const int* __restrict__ decay_flags,
    int nchunks, float lr, float b1, float b2, float eps, float wd,
    float bc1, float bc2, float grad_scale,
    const float* __restrict__ clip_coef) {
  const float clip = *clip_coef;
  for (int c = blockIdx.x; c < nchunks; c += gridDim.x) {
    const long long s = starts[c];
    const long long e = ends[c];
    const float wdc = decay_flags[c] ? wd : 0.f;
    for (long long i = s + threadIdx.x; i < e; i += ADAMW_BLOCK) {
      float g = IS_BF16 ? bf2f(((const short*)grads)[i]) : ((const float*)grads)[i];
      g *= grad_scale * clip;
      float m = exp_avg[i] = b1 * exp_avg[i] + (1.f - b1) * g;
      float v = exp_avg_sq[i] = b2 * exp_avg_sq[i] + (1.f - b2) * g * g;
      float mhat = m / bc1;
      float vhat = v / bc2;
      float p = master[i];
      p -= lr * (mhat / (sqrtf(vhat) + eps) + wdc * p);
      master[i] = p;
      if (IS_BF16) ((short*)params)[i] = f2bf(p);
      else ((float*)params)[i] = p;
    }
  }
}

extern "C" {

void fused_adamw_launch(float* master, void* params, const void* grads,
                        float* exp_avg, float* exp_avg_sq,
                        const long long* starts, const long long* ends,
                        const int* decay_flags, int nchunks, float lr,
                        float b1, float b2, float eps, float wd, int step,
                        float grad_scale, const float* clip_coef,
                        bool is_bf16, hipStream_t stream) {
  float bc1 = 1.f - powf(b1, (float)step);
  float bc2 = 1.f - powf(b2, (float)step);
  int grid = nchunks < 2048 ? nchunks : 2048;
  if (is_bf16)
    fused_adamw_kernel<true><<<grid, ADAMW_BLOCK, 0, stream>>>(
        master, params, grads, exp_avg, exp_avg_sq, starts, ends, decay_flags,
        nchunks, lr, b1, b2, eps, wd, bc1, bc2, grad_scale, clip_coef);
  else
    fused_adamw_kernel<false><<<grid, ADAMW_BLOCK, 0, stream>>>(
        master, params, grads, exp_avg, exp_avg_sq, starts, ends, decay_flags,
        nchunks, lr, b1, b2, eps, wd, bc1, bc2, grad_scale, clip_coef);
}

}  // extern "C"
