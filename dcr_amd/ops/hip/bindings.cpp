// PyTorch bindings for the dcr_amd HIP kernel library (MI355X / gfx950).
// Host-only TU: dispatches dtypes + shapes and calls launchers from the
// .hip TUs. Fails loudly on unsupported layouts rather than silently
// falling back (bench validity: the HIP path must be the one that runs).

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include "dcr_launchers.h"

using namespace dcr;

namespace {

DType dtype_of(const at::Tensor& t) {
  switch (t.scalar_type()) {
    case at::kFloat: return DT_F32;
    case at::kHalf: return DT_F16;
    case at::kBFloat16: return DT_BF16;
    default: TORCH_CHECK(false, "dcr_hip: unsupported dtype ", t.scalar_type());
  }
}

hipStream_t cur_stream() {
  return at::cuda::getCurrentCUDAStream().stream();
}

at::Tensor as_f32(const at::Tensor& t) {
  return t.scalar_type() == at::kFloat ? t.contiguous()
                                       : t.to(at::kFloat).contiguous();
}

// weights may stay in the activation dtype (pure-bf16 training: avoids a
// cast kernel per norm call) or be fp32 (autocast); anything else -> fp32.
at::Tensor norm_weight(const at::Tensor& w, const at::Tensor& x, bool& w_f32) {
  if (w.scalar_type() == at::kFloat) {
    w_f32 = true;
    return w.contiguous();
  }
  if (w.scalar_type() == x.scalar_type()) {
    w_f32 = false;
    return w.contiguous();
  }
  w_f32 = true;
  return w.to(at::kFloat).contiguous();
}

}  // namespace

// ---------------------------------------------------------------- GroupNorm
std::vector<at::Tensor> groupnorm_silu_fwd(at::Tensor x, at::Tensor w, at::Tensor b,
                                           int64_t groups, double eps, bool silu) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous(), "gn: x must be contiguous CUDA");
  TORCH_CHECK(x.dim() >= 2, "gn: x must be [N, C, ...]");
  const int64_t N = x.size(0), C = x.size(1);
  const int64_t HW = x.numel() / (N * C);
  TORCH_CHECK(C % groups == 0, "gn: C % groups != 0");
  bool wf32;
  auto wf = norm_weight(w, x, wf32);
  bool bf32;
  auto bf = norm_weight(b, x, bf32);
  TORCH_CHECK(wf32 == bf32);
  auto y = at::empty_like(x);
  auto mean = at::empty({N * groups}, x.options().dtype(at::kFloat));
  auto rstd = at::empty_like(mean);
  gn_fwd_launch(dtype_of(x), x.data_ptr(), wf.data_ptr(), bf.data_ptr(), wf32,
                y.data_ptr(), mean.data_ptr<float>(), rstd.data_ptr<float>(),
                (int)(N * groups), (int)groups, (int)(C / groups), (int)HW,
                (float)eps, silu, cur_stream());
  return {y, mean, rstd};
}

std::vector<at::Tensor> groupnorm_silu_bwd(at::Tensor dy, at::Tensor x, at::Tensor w,
                                           at::Tensor b, at::Tensor mean, at::Tensor rstd,
                                           int64_t groups, bool silu) {
  TORCH_CHECK(dy.is_contiguous() && x.is_contiguous());
  const int64_t N = x.size(0), C = x.size(1);
  const int64_t HW = x.numel() / (N * C);
  bool wf32;
  auto wf = norm_weight(w, x, wf32);
  bool bf32;
  auto bf = norm_weight(b, x, bf32);
  auto dx = at::empty_like(x);
  auto dwdb = at::zeros({2 * C}, x.options().dtype(at::kFloat));
  auto dw = dwdb.narrow(0, 0, C);
  auto db = dwdb.narrow(0, C, C);
  gn_bwd_launch(dtype_of(x), dy.data_ptr(), x.data_ptr(), wf.data_ptr(),
                bf.data_ptr(), wf32, mean.data_ptr<float>(), rstd.data_ptr<float>(),
                dx.data_ptr(), dw.data_ptr<float>(), db.data_ptr<float>(),
                (int)(N * groups), (int)groups, (int)(C / groups), (int)HW,
                silu, cur_stream());
  return {dx, dw.to(w.scalar_type()), db.to(b.scalar_type())};
}

// channels_last GroupNorm: x is NHWC-contiguous, passed as [N, C, H, W]
// logical sizes with channels_last memory format.
std::vector<at::Tensor> groupnorm_silu_nhwc_fwd(at::Tensor x, at::Tensor w,
                                                at::Tensor b, int64_t groups,
                                                double eps, bool silu) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 &&
              x.is_contiguous(at::MemoryFormat::ChannelsLast),
              "gn_nhwc: channels_last 4D expected");
  const int64_t N = x.size(0), C = x.size(1);
  const int64_t R = x.size(2) * x.size(3);
  TORCH_CHECK(C % groups == 0 && C % 4 == 0);
  bool wf32;
  auto wf = norm_weight(w, x, wf32);
  bool bf32;
  auto bf = norm_weight(b, x, bf32);
  TORCH_CHECK(wf32 == bf32);
  auto y = at::empty_like(x);  // preserves channels_last
  const int64_t chunks = gn_nhwc_chunks((int)N, (int)R);
  // per-chunk partial slab, plain stores -> at::empty (no fill kernel)
  auto ws = at::empty({chunks * N * groups * 2},
                      x.options().dtype(at::kFloat));
  auto mean = at::empty({N * groups}, x.options().dtype(at::kFloat));
  auto rstd = at::empty_like(mean);
  gn_nhwc_fwd_launch(dtype_of(x), x.data_ptr(), wf.data_ptr(), bf.data_ptr(),
                     wf32, y.data_ptr(), ws.data_ptr<float>(),
                     mean.data_ptr<float>(), rstd.data_ptr<float>(), (int)N,
                     (int)R, (int)C, (int)groups, (float)eps, silu, cur_stream());
  return {y, mean, rstd};
}

std::vector<at::Tensor> groupnorm_silu_nhwc_bwd(at::Tensor dy, at::Tensor x,
                                                at::Tensor w, at::Tensor b,
                                                at::Tensor mean, at::Tensor rstd,
                                                int64_t groups, bool silu) {
  TORCH_CHECK(dy.is_contiguous(at::MemoryFormat::ChannelsLast) &&
              x.is_contiguous(at::MemoryFormat::ChannelsLast));
  const int64_t N = x.size(0), C = x.size(1);
  const int64_t R = x.size(2) * x.size(3);
  bool wf32;
  auto wf = norm_weight(w, x, wf32);
  bool bf32;
  auto bf = norm_weight(b, x, bf32);
  auto dx = at::empty_like(x);
  const int64_t chunks = gn_nhwc_chunks((int)N, (int)R);
  // [chunk-partial group sums | chunk-partial dw/db | finalized groups]
  // — every slot plain-stored, so no zero-init kernel; dw/db are the
  // finalize kernel's outputs
  auto scratch = at::empty(
      {chunks * N * groups * 2 + chunks * N * 2 * C + N * groups * 2 + 2 * C},
      x.options().dtype(at::kFloat));
  auto ws = scratch.narrow(0, 0, chunks * N * groups * 2);
  auto dw = scratch.narrow(
      0, chunks * N * groups * 2 + chunks * N * 2 * C + N * groups * 2, C);
  auto db = scratch.narrow(
      0, chunks * N * groups * 2 + chunks * N * 2 * C + N * groups * 2 + C, C);
  gn_nhwc_bwd_launch(dtype_of(x), dy.data_ptr(), x.data_ptr(),
                     wf.data_ptr(), bf.data_ptr(), wf32,
                     mean.data_ptr<float>(), rstd.data_ptr<float>(),
                     ws.data_ptr<float>(), dx.data_ptr(), dw.data_ptr<float>(),
                     db.data_ptr<float>(), (int)N, (int)R, (int)C, (int)groups,
                     silu, cur_stream());
  return {dx, dw.to(w.scalar_type()), db.to(b.scalar_type())};
}

// ---------------------------------------------------------------- LayerNorm
std::vector<at::Tensor> layernorm_fwd(at::Tensor x, at::Tensor w, at::Tensor b,
                                      double eps) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous());
  const int64_t Nd = x.size(-1);
  const int64_t M = x.numel() / Nd;
  bool wf32;
  auto wf = norm_weight(w, x, wf32);
  bool bf32;
  auto bf = norm_weight(b, x, bf32);
  TORCH_CHECK(wf32 == bf32);
  auto y = at::empty_like(x);
  auto mean = at::empty({M}, x.options().dtype(at::kFloat));
  auto rstd = at::empty_like(mean);
  ln_fwd_launch(dtype_of(x), x.data_ptr(), wf.data_ptr(), bf.data_ptr(), wf32,
                y.data_ptr(), mean.data_ptr<float>(), rstd.data_ptr<float>(),
                M, (int)Nd, (float)eps, cur_stream());
  return {y, mean, rstd};
}

std::vector<at::Tensor> layernorm_bwd(at::Tensor dy, at::Tensor x, at::Tensor w,
                                      at::Tensor mean, at::Tensor rstd) {
  TORCH_CHECK(dy.is_contiguous() && x.is_contiguous());
  const int64_t Nd = x.size(-1);
  const int64_t M = x.numel() / Nd;
  TORCH_CHECK(Nd <= 16384, "ln_bwd: N too large for LDS accumulator");
  bool wf32;
  auto wf = norm_weight(w, x, wf32);
  auto dx = at::empty_like(x);
  auto dwdb = at::zeros({2 * Nd}, x.options().dtype(at::kFloat));
  auto dw = dwdb.narrow(0, 0, Nd);
  auto db = dwdb.narrow(0, Nd, Nd);
  ln_bwd_launch(dtype_of(x), dy.data_ptr(), x.data_ptr(), wf.data_ptr(), wf32,
                mean.data_ptr<float>(), rstd.data_ptr<float>(), dx.data_ptr(),
                dw.data_ptr<float>(), db.data_ptr<float>(), M, (int)Nd,
                cur_stream());
  return {dx, dw.to(w.scalar_type()), db.to(w.scalar_type())};
}

// ---------------------------------------------------------------- GEGLU
at::Tensor geglu_fwd(at::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous());
  const int64_t twoN = x.size(-1);
  TORCH_CHECK(twoN % 2 == 0);
  const int64_t N = twoN / 2;
  TORCH_CHECK(N % 4 == 0, "geglu: inner dim must be divisible by 4");
  const int64_t M = x.numel() / twoN;
  auto sizes = x.sizes().vec();
  sizes.back() = N;
  auto y = at::empty(sizes, x.options());
  geglu_fwd_launch(dtype_of(x), x.data_ptr(), y.data_ptr(), M, N, cur_stream());
  return y;
}

at::Tensor geglu_bwd(at::Tensor dy, at::Tensor x) {
  TORCH_CHECK(dy.is_contiguous() && x.is_contiguous());
  const int64_t twoN = x.size(-1);
  const int64_t N = twoN / 2;
  const int64_t M = x.numel() / twoN;
  auto dx = at::empty_like(x);
  geglu_bwd_launch(dtype_of(x), dy.data_ptr(), x.data_ptr(), dx.data_ptr(), M, N,
                   cur_stream());
  return dx;
}

// ---------------------------------------------------------------- AdamW
void adamw_step(at::Tensor p, at::Tensor g, at::Tensor m, at::Tensor v,
                double lr, double beta1, double beta2, double eps, double wd,
                int64_t step) {
  TORCH_CHECK(p.is_cuda() && p.scalar_type() == at::kFloat,
              "adamw: fp32 flat params expected");
  TORCH_CHECK(p.is_contiguous() && g.is_contiguous() && m.is_contiguous() &&
              v.is_contiguous());
  adamw_launch(p.data_ptr<float>(), g.data_ptr<float>(), m.data_ptr<float>(),
               v.data_ptr<float>(), p.numel(), (float)lr, (float)beta1,
               (float)beta2, (float)eps, (float)wd, step, cur_stream());
}

void adamw_step_bf16(at::Tensor p, at::Tensor g, at::Tensor master,
                     at::Tensor m, at::Tensor v, double lr, double beta1,
                     double beta2, double eps, double wd, int64_t step) {
  TORCH_CHECK(p.is_cuda() && p.scalar_type() == at::kBFloat16);
  TORCH_CHECK(master.scalar_type() == at::kFloat);
  TORCH_CHECK(p.is_contiguous() && g.is_contiguous() && master.is_contiguous());
  adamw_bf16_launch(p.data_ptr(), g.data_ptr(), master.data_ptr<float>(),
                    m.data_ptr<float>(), v.data_ptr<float>(), p.numel(),
                    (float)lr, (float)beta1, (float)beta2, (float)eps,
                    (float)wd, step, cur_stream());
}

// device-state AdamW (round-2 draft): hyper = float[8] cuda tensor
// {lr, b1^t, b2^t, inv_bc1, inv_bc2, clip_coef, gnorm_sq, step};
// init {lr, 1, 1, 1, 1, 1, 0, 0}. The 3-kernel sequence is stream-ordered
// and hipGraph-capturable; update hyper[0] (lr) from the host between
// replays. Gated validation: DCR_DEV_ADAMW=1 tests.
void adamw_step_dev(at::Tensor p, at::Tensor g, c10::optional<at::Tensor> master,
                    at::Tensor m, at::Tensor v, at::Tensor hyper, double beta1,
                    double beta2, double eps, double wd, double max_norm) {
  TORCH_CHECK(hyper.is_cuda() && hyper.scalar_type() == at::kFloat &&
              hyper.numel() == 8 && hyper.is_contiguous());
  TORCH_CHECK(p.is_contiguous() && g.is_contiguous());
  const bool bf16 = p.scalar_type() == at::kBFloat16;
  float* mp = nullptr;
  if (bf16) {
    TORCH_CHECK(master.has_value() && master->scalar_type() == at::kFloat,
                "bf16 device-state adamw needs the fp32 master");
    mp = master->data_ptr<float>();
  } else {
    TORCH_CHECK(p.scalar_type() == at::kFloat);
  }
  adamw_dev_launch(bf16 ? 1 : 0, p.data_ptr(), g.data_ptr(), mp,
                   m.data_ptr<float>(), v.data_ptr<float>(), p.numel(),
                   (float)beta1, (float)beta2, (float)eps, (float)wd,
                   (float)max_norm, hyper.data_ptr<float>(), cur_stream());
}

// ------------------------------------------------------------- scheduler math
static at::Tensor sched_common(int mode, at::Tensor a, at::Tensor b, at::Tensor ac,
                               at::Tensor t) {
  TORCH_CHECK(a.is_cuda() && a.is_contiguous() && b.is_contiguous());
  TORCH_CHECK(ac.scalar_type() == at::kFloat && t.scalar_type() == at::kLong);
  const int64_t B = t.size(0);
  const int64_t per = a.numel() / B;
  TORCH_CHECK(per % 4 == 0, "sched: per-sample numel must be divisible by 4");
  auto out = at::empty_like(a);
  sched_launch(dtype_of(a), mode, a.data_ptr(), b.data_ptr(), ac.data_ptr<float>(),
               t.data_ptr<int64_t>(), out.data_ptr(), per, a.numel(), cur_stream());
  return out;
}

at::Tensor add_noise(at::Tensor x0, at::Tensor noise, at::Tensor ac, at::Tensor t) {
  return sched_common(0, x0, noise, ac, t);
}

at::Tensor get_velocity(at::Tensor x0, at::Tensor noise, at::Tensor ac, at::Tensor t) {
  return sched_common(1, x0, noise, ac, t);
}

// elementwise over any DENSE layout (NCHW or channels_last) as long as
// every operand shares the same strides — the kernel walks raw memory.
static bool same_dense(const at::Tensor& a, const at::Tensor& b) {
  return a.is_non_overlapping_and_dense() && b.is_non_overlapping_and_dense() &&
         a.strides() == b.strides();
}

at::Tensor lincomb(at::Tensor X, at::Tensor Y,
                   c10::optional<at::Tensor> Z, double a, double b, double c) {
  TORCH_CHECK(X.is_cuda());
  TORCH_CHECK(same_dense(X, Y), "lincomb: X/Y layout mismatch");
  TORCH_CHECK(X.numel() % 4 == 0);
  auto out = at::empty_like(X);  // preserves layout
  const void* zp = nullptr;
  at::Tensor Zc;
  if (Z.has_value()) {
    TORCH_CHECK(same_dense(X, *Z), "lincomb: X/Z layout mismatch");
    Zc = *Z;
    zp = Zc.data_ptr();
  }
  lincomb_launch(dtype_of(X), X.data_ptr(), Y.data_ptr(), zp, out.data_ptr(),
                 (float)a, (float)b, (float)c, X.numel(), cur_stream());
  return out;
}

at::Tensor cfg_combine(at::Tensor eu, at::Tensor et, double scale) {
  TORCH_CHECK(eu.is_cuda());
  TORCH_CHECK(same_dense(eu, et), "cfg: layout mismatch");
  TORCH_CHECK(eu.numel() % 4 == 0);
  auto out = at::empty_like(eu);
  cfg_launch(dtype_of(eu), eu.data_ptr(), et.data_ptr(), out.data_ptr(),
             (float)scale, eu.numel(), cur_stream());
  return out;
}

// ---------------------------------------------------------------- attention
// q/k/v: [B, L, H, 64] (transpose-free BLHD layout; pass 3D [BH, L, 64]
// for the packed per-head layout — treated as B'=BH, H=1).
static void attn_dims(const at::Tensor& q, const at::Tensor& k,
                      int64_t& B, int64_t& H, int64_t& Lq, int64_t& Lk) {
  TORCH_CHECK(q.size(-1) == 64 && k.size(-1) == 64, "attn: head_dim 64 only");
  if (q.dim() == 4) {
    B = q.size(0); Lq = q.size(1); H = q.size(2); Lk = k.size(1);
    TORCH_CHECK(k.dim() == 4 && k.size(2) == H, "attn: head mismatch");
  } else {
    TORCH_CHECK(q.dim() == 3);
    B = q.size(0); Lq = q.size(1); H = 1; Lk = k.size(1);
  }
}

std::vector<at::Tensor> attn_fwd(at::Tensor q, at::Tensor k, at::Tensor v,
                                 double scale, bool causal) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == at::kBFloat16,
              "attn: bf16 CUDA only");
  TORCH_CHECK(q.is_contiguous() && k.is_contiguous() && v.is_contiguous());
  int64_t B, H, Lq, Lk;
  attn_dims(q, k, B, H, Lq, Lk);
  auto o = at::empty_like(q);
  auto lse = at::empty({B * H, Lq}, q.options().dtype(at::kFloat));
  attn_fwd_launch(q.data_ptr(), k.data_ptr(), v.data_ptr(), o.data_ptr(),
                  lse.data_ptr<float>(), (int)(B * H), (int)Lq, (int)Lk,
                  (int)H, (float)scale, causal, cur_stream());
  return {o, lse};
}

// generalized head-dim forward (SD-1.4 40/80/160; inference-only — no
// backward: sd_mitigation / diff_inference sampling runs under no_grad)
at::Tensor attn_fwd_gen(at::Tensor q, at::Tensor k, at::Tensor v,
                        double scale, bool causal) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == at::kBFloat16,
              "attn_gen: bf16 CUDA only");
  TORCH_CHECK(q.is_contiguous() && k.is_contiguous() && v.is_contiguous());
  const int64_t D = q.size(-1);
  TORCH_CHECK(D == 40 || D == 80 || D == 160,
              "attn_gen: head_dim 40/80/160 only (64 has the main kernel)");
  int64_t B, H, Lq, Lk;
  if (q.dim() == 4) {
    B = q.size(0); Lq = q.size(1); H = q.size(2); Lk = k.size(1);
    TORCH_CHECK(k.dim() == 4 && k.size(2) == H && k.size(-1) == D);
  } else {
    TORCH_CHECK(q.dim() == 3);
    B = q.size(0); Lq = q.size(1); H = 1; Lk = k.size(1);
  }
  auto o = at::empty_like(q);
  attn_fwd_gen_launch(q.data_ptr(), k.data_ptr(), v.data_ptr(), o.data_ptr(),
                      nullptr, (int)(B * H), (int)Lq, (int)Lk, (int)H, (int)D,
                      (float)scale, causal, cur_stream());
  return o;
}

// round-2 draft: T14 async K/V staging + one barrier per tile + setprio
// (DCR_ATTN_V3=1 gates its tests and dispatch)
std::vector<at::Tensor> attn_fwd_v4(at::Tensor q, at::Tensor k, at::Tensor v,
                                    double scale, bool causal) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == at::kBFloat16,
              "attn: bf16 CUDA only");
  TORCH_CHECK(q.is_contiguous() && k.is_contiguous() && v.is_contiguous());
  int64_t B, H, Lq, Lk;
  attn_dims(q, k, B, H, Lq, Lk);
  auto o = at::empty_like(q);
  auto lse = at::empty({B * H, Lq}, q.options().dtype(at::kFloat));
  attn_fwd_v4_launch(q.data_ptr(), k.data_ptr(), v.data_ptr(), o.data_ptr(),
                     lse.data_ptr<float>(), (int)(B * H), (int)Lq, (int)Lk,
                     (int)H, (float)scale, causal, cur_stream());
  return {o, lse};
}

std::vector<at::Tensor> attn_fwd_v3(at::Tensor q, at::Tensor k, at::Tensor v,
                                    double scale, bool causal) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == at::kBFloat16,
              "attn: bf16 CUDA only");
  TORCH_CHECK(q.is_contiguous() && k.is_contiguous() && v.is_contiguous());
  int64_t B, H, Lq, Lk;
  attn_dims(q, k, B, H, Lq, Lk);
  auto o = at::empty_like(q);
  auto lse = at::empty({B * H, Lq}, q.options().dtype(at::kFloat));
  attn_fwd_v3_launch(q.data_ptr(), k.data_ptr(), v.data_ptr(), o.data_ptr(),
                     lse.data_ptr<float>(), (int)(B * H), (int)Lq, (int)Lk,
                     (int)H, (float)scale, causal, cur_stream());
  return {o, lse};
}

std::vector<at::Tensor> attn_fwd_v2(at::Tensor q, at::Tensor k, at::Tensor v,
                                    double scale, bool causal) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == at::kBFloat16,
              "attn: bf16 CUDA only");
  TORCH_CHECK(q.is_contiguous() && k.is_contiguous() && v.is_contiguous());
  int64_t B, H, Lq, Lk;
  attn_dims(q, k, B, H, Lq, Lk);
  auto o = at::empty_like(q);
  auto lse = at::empty({B * H, Lq}, q.options().dtype(at::kFloat));
  attn_fwd_v2_launch(q.data_ptr(), k.data_ptr(), v.data_ptr(), o.data_ptr(),
                     lse.data_ptr<float>(), (int)(B * H), (int)Lq, (int)Lk,
                     (int)H, (float)scale, causal, cur_stream());
  return {o, lse};
}

std::vector<at::Tensor> attn_bwd(at::Tensor q, at::Tensor k, at::Tensor v,
                                 at::Tensor o, at::Tensor dO, at::Tensor lse,
                                 double scale, bool causal) {
  int64_t B, H, Lq, Lk;
  attn_dims(q, k, B, H, Lq, Lk);
  auto dQ = at::empty_like(q);
  auto dK = at::empty_like(k);
  auto dV = at::empty_like(v);
  auto delta = at::empty({B * H, Lq}, q.options().dtype(at::kFloat));
  attn_bwd_launch(q.data_ptr(), k.data_ptr(), v.data_ptr(), o.data_ptr(),
                  dO.contiguous().data_ptr(), lse.data_ptr<float>(),
                  delta.data_ptr<float>(), dQ.data_ptr(), dK.data_ptr(),
                  dV.data_ptr(), (int)(B * H), (int)Lq, (int)Lk, (int)H,
                  (float)scale, causal, cur_stream());
  return {dQ, dK, dV};
}

// backward v4 draft (DCR_ATTN_BWD_V4): swapped-operand schedule
std::vector<at::Tensor> attn_bwd_v4(at::Tensor q, at::Tensor k, at::Tensor v,
                                 at::Tensor o, at::Tensor dO, at::Tensor lse,
                                 double scale, bool causal) {
  int64_t B, H, Lq, Lk;
  attn_dims(q, k, B, H, Lq, Lk);
  auto dQ = at::empty_like(q);
  auto dK = at::empty_like(k);
  auto dV = at::empty_like(v);
  auto delta = at::empty({B * H, Lq}, q.options().dtype(at::kFloat));
  attn_bwd_v4_launch(q.data_ptr(), k.data_ptr(), v.data_ptr(), o.data_ptr(),
                  dO.contiguous().data_ptr(), lse.data_ptr<float>(),
                  delta.data_ptr<float>(), dQ.data_ptr(), dK.data_ptr(),
                  dV.data_ptr(), (int)(B * H), (int)Lq, (int)Lk, (int)H,
                  (float)scale, causal, cur_stream());
  return {dQ, dK, dV};
}


at::Tensor mfma_probe(at::Tensor A, at::Tensor B) {
  TORCH_CHECK(A.is_cuda() && A.scalar_type() == at::kBFloat16);
  TORCH_CHECK(A.sizes() == at::IntArrayRef({16, 32}) &&
              B.sizes() == at::IntArrayRef({32, 16}));
  auto C = at::zeros({16, 16}, A.options().dtype(at::kFloat));
  mfma_probe_launch(A.contiguous().data_ptr(), B.contiguous().data_ptr(),
                    C.data_ptr<float>(), cur_stream());
  return C;
}

// ---------------------------------------------------------------- conv
at::Tensor conv2d_nhwc_fwd(at::Tensor x, at::Tensor w,
                           c10::optional<at::Tensor> bias, int64_t stride,
                           int64_t pad) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16);
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast) &&
              w.is_contiguous(at::MemoryFormat::ChannelsLast),
              "conv_nhwc: channels_last only");
  const int64_t Nb = x.size(0), C = x.size(1), Hin = x.size(2), Win = x.size(3);
  const int64_t K = w.size(0), R = w.size(2), S = w.size(3);
  TORCH_CHECK(w.size(1) == C && C % 32 == 0 && K % 64 == 0);
  TORCH_CHECK((R == 3 && S == 3) || (R == 1 && S == 1));
  const int64_t P = (Hin + 2 * pad - R) / stride + 1;
  const int64_t Q = (Win + 2 * pad - S) / stride + 1;
  auto y = at::empty({Nb, K, P, Q},
                     x.options().memory_format(at::MemoryFormat::ChannelsLast));
  at::Tensor bf;
  const float* bp = nullptr;
  if (bias.has_value()) {
    bf = bias->to(at::kFloat).contiguous();
    bp = bf.data_ptr<float>();
  }
  conv_nhwc_fwd_launch(x.data_ptr(), w.data_ptr(), bp, y.data_ptr(), (int)Nb,
                       (int)Hin, (int)Win, (int)C, (int)K, (int)P, (int)Q,
                       (int)R, (int)S, (int)stride, (int)pad, cur_stream());
  return y;
}

static at::Tensor conv2d_nhwc_fwd_v2_impl(at::Tensor x, at::Tensor w,
                                          c10::optional<at::Tensor> bias,
                                          int64_t stride, int64_t pad,
                                          c10::optional<at::Tensor> res,
                                          c10::optional<at::Tensor> temb) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16);
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast) &&
              w.is_contiguous(at::MemoryFormat::ChannelsLast));
  const int64_t Nb = x.size(0), C = x.size(1), Hin = x.size(2), Win = x.size(3);
  const int64_t K = w.size(0), R = w.size(2), S = w.size(3);
  TORCH_CHECK(w.size(1) == C && C % 32 == 0 && K % 64 == 0);
  TORCH_CHECK((R == 3 && S == 3) || (R == 1 && S == 1));
  const int64_t P = (Hin + 2 * pad - R) / stride + 1;
  const int64_t Q = (Win + 2 * pad - S) / stride + 1;
  auto y = at::empty({Nb, K, P, Q},
                     x.options().memory_format(at::MemoryFormat::ChannelsLast));
  at::Tensor bf;
  const float* bp = nullptr;
  if (bias.has_value()) {
    bf = bias->to(at::kFloat).contiguous();
    bp = bf.data_ptr<float>();
  }
  // split-K when the (pixels/128) x (K/128) grid starves the 256 CUs
  const int64_t NPQ = Nb * P * Q;
  int64_t blocks = ((NPQ + 127) / 128) * ((K + 127) / 128);
  int64_t nsteps = C * R * S / ((C % 64 == 0) ? 64 : 32);
  int splitz = 1;
  if (blocks < 384) {
    splitz = (int)std::min<int64_t>({(384 + blocks - 1) / blocks, nsteps, 16});
    if (splitz < 1) splitz = 1;
  }
  at::Tensor ws;
  float* wsp = nullptr;
  if (splitz > 1) {
    ws = at::zeros({NPQ * K}, x.options().dtype(at::kFloat));
    wsp = ws.data_ptr<float>();
  }
  const void* rp = nullptr;
  const void* tp = nullptr;
  if (res.has_value()) {
    TORCH_CHECK(res->scalar_type() == at::kBFloat16 &&
                res->is_contiguous(at::MemoryFormat::ChannelsLast) &&
                res->sizes() == y.sizes(), "conv res: NHWC bf16 same shape");
    rp = res->data_ptr();
  }
  if (temb.has_value()) {
    TORCH_CHECK(temb->scalar_type() == at::kBFloat16 &&
                temb->is_contiguous() && temb->dim() == 2 &&
                temb->size(0) == Nb && temb->size(1) == K,
                "conv temb: contiguous bf16 [N, K]");
    tp = temb->data_ptr();
  }
  conv_nhwc_fwd_v2_launch(x.data_ptr(), w.data_ptr(), bp, y.data_ptr(), wsp,
         splitz, rp, tp, (int)Nb, (int)Hin, (int)Win, (int)C, (int)K,
         (int)P, (int)Q, (int)R, (int)S, (int)stride,
         (int)pad, cur_stream());
  return y;
}

at::Tensor conv2d_nhwc_fwd_v2(at::Tensor x, at::Tensor w,
                              c10::optional<at::Tensor> bias, int64_t stride,
                              int64_t pad,
                              c10::optional<at::Tensor> res = c10::nullopt,
                              c10::optional<at::Tensor> temb = c10::nullopt) {
  return conv2d_nhwc_fwd_v2_impl(x, w, bias, stride, pad, res, temb);
}

// conv backward drafts (round-2; validated before any dispatch)
std::vector<at::Tensor> conv2d_nhwc_bwd(at::Tensor dy, at::Tensor x,
                                        at::Tensor w, int64_t stride,
                                        int64_t pad) {
  TORCH_CHECK(dy.is_cuda() && dy.scalar_type() == at::kBFloat16);
  TORCH_CHECK(dy.is_contiguous(at::MemoryFormat::ChannelsLast) &&
              x.is_contiguous(at::MemoryFormat::ChannelsLast) &&
              w.is_contiguous(at::MemoryFormat::ChannelsLast));
  const int64_t Nb = x.size(0), C = x.size(1), Hin = x.size(2), Win = x.size(3);
  const int64_t K = w.size(0), R = w.size(2), S = w.size(3);
  TORCH_CHECK(C % 64 == 0 && K % 64 == 0);
  const int64_t P = dy.size(2), Q = dy.size(3);
  const int64_t NPQ = Nb * P * Q;
  int splits = (int)std::min<int64_t>(std::max<int64_t>(
      384 / std::max<int64_t>((K / 64) * (R * S * C / 32), 1), 1), 16);
  auto dw_ws = at::zeros({K * R * S * C}, x.options().dtype(at::kFloat));
  auto dW = at::empty_like(w);
  conv_bwd_weight_launch(dy.data_ptr(), x.data_ptr(), dw_ws.data_ptr<float>(),
                         (int)Nb, (int)Hin, (int)Win, (int)C, (int)K, (int)P,
                         (int)Q, (int)R, (int)S, (int)stride, (int)pad,
                         splits, dW.data_ptr(), cur_stream());
  auto dx = at::empty_like(x);
  conv_bwd_data_launch(dy.data_ptr(), w.data_ptr(), dx.data_ptr(), (int)Nb,
                       (int)Hin, (int)Win, (int)C, (int)K, (int)P, (int)Q,
                       (int)R, (int)S, (int)stride, (int)pad, cur_stream());
  auto db = at::zeros({K}, x.options().dtype(at::kFloat));
  conv_bias_grad_launch(dy.data_ptr(), db.data_ptr<float>(), NPQ, (int)K,
                        cur_stream());
  return {dx, dW, db};
}

// ---------------------------------------------------------------- GEMM
// C[M,N] = sum_k A(m,k)*B(n,k) (+bias[n]); ta/tb: operand memory is
// [K][rows] (k-strided) instead of [rows][K]. want_dbias (wgrad): also
// return the column-sum of A's underlying dy (fused into staging).
std::vector<at::Tensor> gemm_bf16(at::Tensor A, at::Tensor B,
                                  c10::optional<at::Tensor> bias,
                                  bool ta, bool tb, bool want_dbias) {
  TORCH_CHECK(A.is_cuda() && A.scalar_type() == at::kBFloat16 &&
              B.scalar_type() == at::kBFloat16, "gemm_bf16: bf16 CUDA only");
  TORCH_CHECK(A.is_contiguous() && B.is_contiguous());
  TORCH_CHECK(A.dim() == 2 && B.dim() == 2);
  const int64_t M = ta ? A.size(1) : A.size(0);
  const int64_t K = ta ? A.size(0) : A.size(1);
  const int64_t N = tb ? B.size(1) : B.size(0);
  TORCH_CHECK((tb ? B.size(0) : B.size(1)) == K, "gemm_bf16: K mismatch");
  TORCH_CHECK(K % 8 == 0, "gemm_bf16: K % 8 != 0");
  auto C = at::empty({M, N}, A.options());
  at::Tensor bf;
  const float* bp = nullptr;
  if (bias.has_value()) {
    bf = bias->to(at::kFloat).contiguous();
    bp = bf.data_ptr<float>();
  }
  // split the contraction when the tile grid starves the 256 CUs
  const int64_t blocks = ((M + 127) / 128) * ((N + 127) / 128);
  const int64_t nsteps = (K + 63) / 64;
  int splitz = 1;
  if (blocks < 384)
    splitz = (int)std::min<int64_t>({(384 + blocks - 1) / blocks, nsteps, 16});
  if (splitz < 1) splitz = 1;
  at::Tensor ws;
  float* wsp = nullptr;
  if (splitz > 1) {
    ws = at::zeros({M * N}, A.options().dtype(at::kFloat));
    wsp = ws.data_ptr<float>();
  }
  at::Tensor db;
  float* dbp = nullptr;
  if (want_dbias) {
    TORCH_CHECK(ta, "gemm_bf16: dbias fusion needs ta (wgrad layout)");
    db = at::zeros({M}, A.options().dtype(at::kFloat));
    dbp = db.data_ptr<float>();
  }
  gemm_bf16_launch(A.data_ptr(), B.data_ptr(), bp, C.data_ptr(), wsp, dbp,
                   M, N, (int)K, ta ? 1 : 0, tb ? 1 : 0, want_dbias ? 1 : 0,
                   splitz, cur_stream());
  if (want_dbias) return {C, db};
  return {C};
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, mod) {
  mod.def("gemm_bf16", &gemm_bf16);
  mod.def("conv2d_nhwc_bwd", &conv2d_nhwc_bwd);
  mod.def("conv2d_nhwc_fwd", &conv2d_nhwc_fwd);
  mod.def("conv2d_nhwc_fwd_v2", &conv2d_nhwc_fwd_v2,
          pybind11::arg("x"), pybind11::arg("w"), pybind11::arg("bias"),
          pybind11::arg("stride"), pybind11::arg("pad"),
          pybind11::arg("res") = c10::nullopt,
          pybind11::arg("temb") = c10::nullopt);
  mod.def("attn_fwd", &attn_fwd);
  mod.def("attn_fwd_gen", &attn_fwd_gen);
  mod.def("attn_fwd_v2", &attn_fwd_v2);
  mod.def("attn_fwd_v3", &attn_fwd_v3);
  mod.def("attn_fwd_v4", &attn_fwd_v4);
  mod.def("attn_bwd", &attn_bwd);
  mod.def("attn_bwd_v4", &attn_bwd_v4);
  mod.def("mfma_probe", &mfma_probe);
  mod.def("groupnorm_silu_fwd", &groupnorm_silu_fwd);
  mod.def("groupnorm_silu_bwd", &groupnorm_silu_bwd);
  mod.def("groupnorm_silu_nhwc_fwd", &groupnorm_silu_nhwc_fwd);
  mod.def("groupnorm_silu_nhwc_bwd", &groupnorm_silu_nhwc_bwd);
  mod.def("layernorm_fwd", &layernorm_fwd);
  mod.def("layernorm_bwd", &layernorm_bwd);
  mod.def("geglu_fwd", &geglu_fwd);
  mod.def("geglu_bwd", &geglu_bwd);
  mod.def("adamw_step", &adamw_step);
  mod.def("adamw_step_bf16", &adamw_step_bf16);
  mod.def("adamw_step_dev", &adamw_step_dev);
  mod.def("add_noise", &add_noise);
  mod.def("get_velocity", &get_velocity);
  mod.def("cfg_combine", &cfg_combine);
  mod.def("lincomb", &lincomb);
}
