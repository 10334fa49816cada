// Host-side launcher API for the dcr_amd HIP kernels.
// bindings.cpp (built by the host compiler) calls these; definitions live
// in the .hip TUs compiled by hipcc for gfx950.
#pragma once

#include <hip/hip_runtime.h>

namespace dcr {

enum DType { DT_F32 = 0, DT_F16 = 1, DT_BF16 = 2 };

// norms.hip (w_f32: weights/bias are fp32; else same dtype as x)
void gn_fwd_launch(DType dt, const void* x, const void* w, const void* b,
                   bool w_f32, void* y, float* mean, float* rstd, int NG,
                   int G, int Cg, int HW, float eps, bool silu, hipStream_t s);
void gn_bwd_launch(DType dt, const void* dy, const void* x, const void* w,
                   const void* b, bool w_f32, const float* mean,
                   const float* rstd, void* dx, float* dw, float* db, int NG,
                   int G, int Cg, int HW, bool silu, hipStream_t s);
void ln_fwd_launch(DType dt, const void* x, const void* w, const void* b,
                   bool w_f32, void* y, float* mean, float* rstd, long M,
                   int N, float eps, hipStream_t s);
void ln_bwd_launch(DType dt, const void* dy, const void* x, const void* w,
                   bool w_f32, const float* mean, const float* rstd, void* dx,
                   float* dw, float* db, long M, int N, hipStream_t s);

// norms_nhwc.hip (channels_last GroupNorm; w_f32 as above).
// Workspaces are per-chunk partial slabs (plain stores, deterministic,
// no zero-init needed): fwd ws = chunks*N*G*2 floats; bwd ws =
// chunks*N*G*2 + chunks*N*2C + N*G*2 floats. chunks = gn_nhwc_chunks().
int gn_nhwc_chunks(int N, int R);
void gn_nhwc_fwd_launch(DType dt, const void* x, const void* w, const void* b,
                        bool w_f32, void* y, float* ws, float* mean, float* rstd,
                        int N, int R, int C, int G, float eps, bool silu,
                        hipStream_t s);
void gn_nhwc_bwd_launch(DType dt, const void* dy, const void* x, const void* w,
                        const void* b, bool w_f32, const float* mean,
                        const float* rstd, float* ws, void* dx, float* dw,
                        float* db, int N, int R, int C, int G, bool silu,
                        hipStream_t s);

// elementwise.hip
void geglu_fwd_launch(DType dt, const void* x, void* y, long M, long N,
                      hipStream_t s);
void geglu_bwd_launch(DType dt, const void* dy, const void* x, void* dx,
                      long M, long N, hipStream_t s);
void adamw_launch(float* p, const float* g, float* m, float* v, long n,
                  float lr, float b1, float b2, float eps, float wd, long step,
                  hipStream_t s);
void adamw_bf16_launch(void* p, const void* g, float* master, float* m,
                       float* v, long n, float lr, float b1, float b2,
                       float eps, float wd, long step, hipStream_t s);
// device-state AdamW (round-2 draft; hipGraph-capturable step)
void adamw_dev_launch(int bf16, void* p, const void* g, float* master,
                      float* m, float* v, long n, float b1, float b2,
                      float eps, float wd, float max_norm, float* hyper,
                      hipStream_t s);
void sched_launch(DType dt, int mode, const void* x0, const void* noise,
                  const float* ac, const long* t, void* out, long per_sample,
                  long total, hipStream_t s);
void cfg_launch(DType dt, const void* eu, const void* et, void* out, float s_,
                long total, hipStream_t s);
void lincomb_launch(DType dt, const void* X, const void* Y, const void* Z,
                    void* out, float a, float b, float c, long total,
                    hipStream_t s);

// attention.hip (bf16, head_dim 64)
void attn_fwd_launch(const void* q, const void* k, const void* v, void* o,
                     float* lse, int BH, int Lq, int Lk, int H, float scale,
                     bool causal, hipStream_t s);
// v2: masked-tail MFMA skip (round-2 draft; not dispatched)
void attn_fwd_v2_launch(const void* q, const void* k, const void* v, void* o,
                        float* lse, int BH, int Lq, int Lk, int H, float scale,
                        bool causal, hipStream_t s);
// v3: T14 async-stage + one barrier/tile + setprio (round-2 draft)
void attn_fwd_v3_launch(const void* q, const void* k, const void* v, void* o,
                        float* lse, int BH, int Lq, int Lk, int H, float scale,
                        bool causal, hipStream_t s);
// v4: swapped-QK^T in-register-softmax schedule (round-2)
void attn_fwd_v4_launch(const void* q, const void* k, const void* v, void* o,
                        float* lse, int BH, int Lq, int Lk, int H, float scale,
                        bool causal, hipStream_t s);
// generalized head-dim forward (SD-1.4 40/80/160; inference-only)
void attn_fwd_gen_launch(const void* q, const void* k, const void* v, void* o,
                         float* lse, int BH, int Lq, int Lk, int H, int D,
                         float scale, bool causal, hipStream_t s);
void attn_bwd_launch(const void* q, const void* k, const void* v,
                     const void* o, const void* dO, const float* lse,
                     float* delta, void* dQ, void* dK, void* dV, int BH,
                     int Lq, int Lk, int H, float scale, bool causal,
                     hipStream_t s);
// backward v4 (swapped-operand schedule; DCR_ATTN_BWD_V4 draft)
void attn_bwd_v4_launch(const void* q, const void* k, const void* v,
                        const void* o, const void* dO, const float* lse,
                        float* delta, void* dQ, void* dK, void* dV, int BH,
                        int Lq, int Lk, int H, float scale, bool causal,
                        hipStream_t s);
void mfma_probe_launch(const void* A, const void* B, float* C, hipStream_t s);

// conv_nhwc.hip (implicit-GEMM conv fwd, opt-in)
void conv_nhwc_fwd_launch(const void* x, const void* w, const float* bias,
                          void* y, int Nb, int Hin, int Win, int C, int K,
                          int P, int Q, int R, int S, int stride, int pad,
                          hipStream_t st);
void conv_nhwc_fwd_v2_launch(const void* x, const void* w, const float* bias,
                             void* y, float* ws, int splitz, const void* res,
                             const void* temb, int Nb, int Hin, int Win, int C,
                             int K, int P, int Q, int R, int S, int stride,
                             int pad, hipStream_t st);
// conv_nhwc_bwd.hip (dgrad + wgrad + fused bias-grad; DCR_NATIVE_CONV_BWD)
void conv_bwd_weight_launch(const void* dy, const void* x, float* dw_ws,
                            int Nb, int Hin, int Win, int C, int K, int P,
                            int Q, int R, int S, int stride, int pad,
                            int splits, void* dW_out, hipStream_t st);
void conv_bwd_data_launch(const void* dy, const void* w, void* dx, int Nb,
                          int Hin, int Win, int C, int K, int P, int Q, int R,
                          int S, int stride, int pad, hipStream_t st);
void conv_bias_grad_launch(const void* dy, float* db, long NPQ, int K,
                           hipStream_t st);

// gemm.hip — bf16 MFMA GEMM for transformer linears (fwd/dgrad/wgrad)
void gemm_bf16_launch(const void* A, const void* B, const float* bias,
                      void* C, float* ws, float* dbias, long M, long N, int K,
                      int ta, int tb, int want_dbias, int splitz,
                      hipStream_t st);

}  // namespace dcr
