// Launcher declarations for the ProGen CDNA4 kernels.
#pragma once
#include <hip/hip_runtime.h>

extern "C" {

void ln_shift_fwd_launch(const void* x, const void* res, const void* g,
                         void* y, void* s_out, float* mean, float* rstd,
                         int R, int N, int D, float eps, bool shift,
                         bool is_bf16, hipStream_t stream);
void ln_shift_bwd_launch(const void* dy, const void* ds, const void* x,
                         const void* g, const float* mean, const float* rstd,
                         void* dx, float* dw_part, int nblocks, int R, int N,
                         int D, bool shift, bool is_bf16, hipStream_t stream);

void glu_fwd_launch(const void* h, void* y, long long rows, int H,
                    bool is_bf16, hipStream_t stream);
void glu_bwd_launch(const void* dy, const void* h, void* dh, long long rows,
                    int H, bool is_bf16, hipStream_t stream);
void gelu_fwd_launch(const void* h, void* y, long long total_elems,
                     bool is_bf16, hipStream_t stream);
void gelu_bwd_launch(const void* dy, const void* h, void* dh,
                     long long total_elems, bool is_bf16, hipStream_t stream);

void ce_fwd_launch(const void* logits, const long long* targets, float* nll,
                   float* lse, long long R, int V, bool is_bf16,
                   hipStream_t stream);
void ce_bwd_launch(const float* dnll, const void* logits,
                   const long long* targets, const float* lse, void* dlogits,
                   long long R, int V, bool is_bf16, hipStream_t stream);

void grad_sumsq_launch(const void* grads, float* out, long long numel,
                       bool is_bf16, hipStream_t stream);
void fused_adamw_launch(float* master, void* params, const void* grads,
                        float* exp_avg, float* exp_avg_sq,
                        const long long* starts, const long long* ends,
                        const int* decay_flags, int nchunks, float lr,
                        float b1, float b2, float eps, float wd,
                        int* step_dev, float grad_scale,
                        const float* clip_coef, bool is_bf16,
                        long long shard_off, hipStream_t stream);

void colsum_launch(const void* dy, float* out, long long R, int C,
                   hipStream_t stream);
void quant_e4m3_launch(const void* in, void* out, const float* scale,
                       long long numel, hipStream_t stream);
void quant_e4m3_t_launch(const void* in, void* out, const float* scale,
                         int Nr, int Kc, hipStream_t stream);

void rope_qkv_launch(const void* qkv, const float* rsin, const float* rcos,
                     void* qkv_rot, int B, int N, int H, hipStream_t stream);
void attn_fwd_launch(const void* qkv_rot, const void* halo, void* out,
                     float* lse, int B, int N, int H, int wsz,
                     hipStream_t stream);
void attn_bwd_launch(const void* dout, const void* qkv, const void* halo,
                     const float* rsin, const float* rcos, const void* out,
                     const float* lse, float* dacc, float* dlook,
                     float* dhalo, void* dqkv, int B, int N, int H, int wsz,
                     hipStream_t stream);

void sgu_fwd_launch(const void* xa, const void* g_ln, const void* w,
                    const float* bias, void* out, void* gate_out, int B,
                    int N, int D, hipStream_t stream);
void sgu_dgate_launch(const void* t_in, const void* w, void* dg, int B, int N,
                      int D, hipStream_t stream);
void sgu_dw_launch(const void* t_in, const void* g_ln, float* dw,
                   const int* tri_m, const int* tri_k, int ntri, int B, int N,
                   int D, hipStream_t stream);

}  // extern "C"
