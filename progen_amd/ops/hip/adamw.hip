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

// step counter lives ON DEVICE so the kernel computes its own bias
// correction — required for hipGraph capture (a host-baked step would be
// frozen at capture time). step_inc_kernel runs just before on the same
// stream.
// Non-finite-gradient skip (GradScaler semantics): inf/NaN grads make
// the global norm inf (-> clip coef 0) or NaN (-> coef NaN); such a
// step is SKIPPED outright — no moment update, no param update, no
// step-count advance — instead of poisoning the master weights with
// inf*0 = NaN. Healthy steps always have coef in (0, 1], so this is a
// no-op on them. (Also the mitigation for the open graphed-replay
// corruption issue: profiles/r02_graphed_nan_investigation.md.)
__device__ __forceinline__ bool step_ok(float clip) {
  return clip > 0.f && !__builtin_isinf(clip) && !__builtin_isnan(clip);
}

__global__ void step_inc_kernel(int* step, const float* clip_coef) {
  if (threadIdx.x == 0 && blockIdx.x == 0 && step_ok(*clip_coef)) *step += 1;
}

template <bool IS_BF16>
__global__ __launch_bounds__(ADAMW_BLOCK) void fused_adamw_kernel(
    float* __restrict__ master, void* __restrict__ params,
    const void* __restrict__ grads, float* __restrict__ exp_avg,
    float* __restrict__ exp_avg_sq, const long long* __restrict__ starts,
    const long long* __restrict__ ends, const int* __restrict__ decay_flags,
    int nchunks, float lr, float b1, float b2, float eps, float wd,
    const int* __restrict__ step_ptr, float grad_scale,
    const float* __restrict__ clip_coef, long long shard_off) {
  // shard_off: ZeRO-1 — master/exp_avg/exp_avg_sq hold only the rank's
  // [lo, hi) slice of the flat space; chunk tables are pre-clipped to
  // that range and the fp32 state is indexed at (i - shard_off).
  const float clip = *clip_coef;
  if (!step_ok(clip)) return;  // skip the whole update (see step_ok)
  const int step = *step_ptr;
  const float bc1 = 1.f - powf(b1, (float)step);
  const float bc2 = 1.f - powf(b2, (float)step);
  for (int c = blockIdx.x; c < nchunks; c += gridDim.x) {
    const long long s = starts[c];
    const long long e = ends[c];
    const float wdc = decay_flags[c] ? wd : 0.f;
    for (long long i = s + threadIdx.x; i < e; i += ADAMW_BLOCK) {
      float g = IS_BF16 ? bf2f(((const short*)grads)[i]) : ((const float*)grads)[i];
      g *= grad_scale * clip;
      const long long si = i - shard_off;
      float m = exp_avg[si] = b1 * exp_avg[si] + (1.f - b1) * g;
      float v = exp_avg_sq[si] = b2 * exp_avg_sq[si] + (1.f - b2) * g * g;
      float mhat = m / bc1;
      float vhat = v / bc2;
      float p = master[si];
      p -= lr * (mhat / (sqrtf(vhat) + eps) + wdc * p);
      master[si] = p;
      if (IS_BF16) ((short*)params)[i] = f2bf(p);
      else ((float*)params)[i] = p;
    }
  }
}

// grad norm: one-pass sum of squares over the flat grad buffer (fp32
// accumulation), partials combined with a device-scope atomicAdd — the
// clip coefficient then stays on device (no host sync, no fp32 copy of
// the 2.7 GB bf16 grad buffer).
template <bool IS_BF16>
__global__ __launch_bounds__(ADAMW_BLOCK) void grad_sumsq_kernel(
    const void* __restrict__ grads, float* __restrict__ out, long long nv) {
  constexpr int VLEN = IS_BF16 ? 8 : 4;
  float acc = 0.f;
  for (long long i = blockIdx.x * (long long)ADAMW_BLOCK + threadIdx.x;
       i < nv; i += (long long)gridDim.x * ADAMW_BLOCK) {
    if (IS_BF16) {
      bf16x8 v = ((const bf16x8*)grads)[i];
#pragma unroll
      for (int j = 0; j < VLEN; ++j) {
        float f = bf2f(((short*)&v)[j]);
        acc += f * f;
      }
    } else {
      f32x4 v = ((const f32x4*)grads)[i];
#pragma unroll
      for (int j = 0; j < VLEN; ++j) acc += ((float*)&v)[j] * ((float*)&v)[j];
    }
  }
  acc = wave_sum(acc);
  __shared__ float red[ADAMW_BLOCK / WAVE];
  const int wid = threadIdx.x / WAVE, lane = threadIdx.x % WAVE;
  if (lane == 0) red[wid] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    float t = 0.f;
#pragma unroll
    for (int w = 0; w < ADAMW_BLOCK / WAVE; ++w) t += red[w];
    atomicAdd(out, t);
  }
}

extern "C" {

void grad_sumsq_launch(const void* grads, float* out, long long numel,
                       bool is_bf16, hipStream_t stream) {
  long long nv = numel / (is_bf16 ? 8 : 4);
  int grid = (int)((nv + ADAMW_BLOCK - 1) / ADAMW_BLOCK);
  if (grid > 2048) grid = 2048;
  if (is_bf16)
    grad_sumsq_kernel<true><<<grid, ADAMW_BLOCK, 0, stream>>>(grads, out, nv);
  else
    grad_sumsq_kernel<false><<<grid, ADAMW_BLOCK, 0, stream>>>(grads, out, nv);
}

void fused_adamw_launch(float* master, void* params, const void* grads,
                        float* exp_avg, float* exp_avg_sq,
                        const long long* starts, const long long* ends,
                        const int* decay_flags, int nchunks, float lr,
                        float b1, float b2, float eps, float wd,
                        int* step_dev, float grad_scale,
                        const float* clip_coef, bool is_bf16,
                        long long shard_off, hipStream_t stream) {
  step_inc_kernel<<<1, 1, 0, stream>>>(step_dev, clip_coef);
  int grid = nchunks < 2048 ? nchunks : 2048;
  if (is_bf16)
    fused_adamw_kernel<true><<<grid, ADAMW_BLOCK, 0, stream>>>(
        master, params, grads, exp_avg, exp_avg_sq, starts, ends, decay_flags,
        nchunks, lr, b1, b2, eps, wd, step_dev, grad_scale, clip_coef,
        shard_off);
  else
    fused_adamw_kernel<false><<<grid, ADAMW_BLOCK, 0, stream>>>(
        master, params, grads, exp_avg, exp_avg_sq, starts, ends, decay_flags,
        nchunks, lr, b1, b2, eps, wd, step_dev, grad_scale, clip_coef,
        shard_off);
}

}  // extern "C"
