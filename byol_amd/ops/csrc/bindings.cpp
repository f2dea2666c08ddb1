// Python bindings for the byol_amd CDNA4 HIP kernels (gfx950).
#include <torch/extension.h>
#include <cstdint>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>

void launch_flat_ema_update(float* mean, const float* x, float decay,
                            const float* decay_dev, int64_t n,
                            hipStream_t stream);
void launch_byol_loss_forward(const float* p1, const float* p2,
                              const float* z1, const float* z2, float* stats,
                              float* loss, int64_t n, int64_t batch,
                              hipStream_t stream);
void launch_byol_loss_backward(const float* p1, const float* p2,
                               const float* z1, const float* z2,
                               const float* stats, const float* grad_out,
                               float* g1, float* g2, int64_t n,
                               int64_t batch, hipStream_t stream);
void launch_bn_stats(const float* x, float* acc, int64_t m, int c,
                     int slot_mask, hipStream_t stream);
void launch_bn_stats_v2(const float* x, float* acc, int64_t m, int c,
                        int slot_mask, int grid, hipStream_t stream);
void launch_bn_bwd_reduce_v2(const float* dy, const float* y, const float* x,
                             const float* mean, const float* invstd,
                             float* red, int64_t m, int c, int relu,
                             int slot_mask, int grid, hipStream_t stream);
void launch_bn_apply_v2(const float* x, const float* residual,
                        const float* mean, const float* invstd,
                        const float* weight, const float* bias, float* y,
                        int64_t m, int c, int relu, int grid,
                        hipStream_t stream);
void launch_bn_bwd_apply_v2(const float* dy, const float* y, const float* x,
                            const float* mean, const float* invstd,
                            const float* weight, const float* red,
                            float* dx, float* dresidual, float inv_count,
                            int64_t m, int c, int relu, int grid,
                            hipStream_t stream);
void launch_bn_reduce_slots(const float* in, float* out, int n2c, int nslots,
                            hipStream_t stream);
void launch_bn_finalize(const float* acc, float* mean, float* invstd,
                        float* running_mean, float* running_var, float count,
                        float eps, float momentum, int c, int update_running,
                        hipStream_t stream);
void launch_bn_apply(const float* x, const float* residual, const float* mean,
                     const float* invstd, const float* weight,
                     const float* bias, float* y, int64_t m, int c, int relu,
                     hipStream_t stream);
void launch_bn_bwd_reduce(const float* dy, const float* y, const float* x,
                          const float* mean, const float* invstd, float* red,
                          int64_t m, int c, int relu, int slot_mask,
                          hipStream_t stream);
void launch_bn_stats_bf16(const uint16_t* x, float* acc, int64_t m, int c,
                          int slot_mask, hipStream_t stream);
void launch_bn_apply_bf16(const uint16_t* x, const uint16_t* residual,
                          const float* mean, const float* invstd,
                          const float* weight, const float* bias,
                          uint16_t* y, int64_t m, int c, int relu,
                          hipStream_t stream);
void launch_bn_bwd_reduce_bf16(const uint16_t* dy, const uint16_t* y,
                               const uint16_t* x, const float* mean,
                               const float* invstd, float* red, int64_t m,
                               int c, int relu, int slot_mask,
                               hipStream_t stream);
void launch_bn_bwd_apply_bf16(const uint16_t* dy, const uint16_t* y,
                              const uint16_t* x, const float* mean,
                              const float* invstd, const float* weight,
                              const float* red, uint16_t* dx,
                              uint16_t* dresidual, float inv_count,
                              int64_t m, int c, int relu,
                              hipStream_t stream);
void launch_bn_bwd_apply(const float* dy, const float* y, const float* x,
                         const float* mean, const float* invstd,
                         const float* weight, const float* red, float* dx,
                         float* dresidual, float inv_count, int64_t m, int c,
                         int relu, hipStream_t stream);
void launch_aug_sample(const float* src, float* dst, float* gray_sum,
                       const float* crop, int b, int hs, int ws, int s,
                       int use_v2, hipStream_t stream);
void launch_aug_color(float* img, const float* gray_sum, const float* cparam,
                      int b, int s, hipStream_t stream);
void launch_aug_blur(const float* img, float* tmp, float* out,
                     const float* sigma, float* wts, int b, int s,
                     int ksize, hipStream_t stream);
void launch_ce_topk_fwd(const float* logits, const int64_t* labels,
                        float* out, float* row_stats, int m, int n,
                        hipStream_t stream);
void launch_ce_bwd(const float* logits, const int64_t* labels,
                   const float* row_stats, const float* grad_out,
                   float* dlogits, int m, int n, hipStream_t stream);
void launch_conv1x1_fwd(const float* x, const float* w, const float* wt,
                        float* y, int64_t m, int k, int n,
                        hipStream_t stream);
void launch_conv1x1_dgrad(const float* dy, const float* w, float* dx,
                          int64_t m, int n, int k, hipStream_t stream);
void launch_conv1x1_wgrad(const float* dy, const float* x, float* dw,
                          int64_t m, int n, int k, hipStream_t stream);
int conv1x1_wgrad_nchunks(int64_t m, int n, int k);
void launch_conv1x1_wgrad_partial(const float* dy, const float* x,
                                  float* partial, float* dw, int64_t m,
                                  int n, int k, hipStream_t stream);
void launch_conv3x3_fwd(const float* x, const float* wp, float* y, int b,
                        int hi, int wi, int ho, int wo, int k, int n,
                        int stride, hipStream_t stream);
void launch_pad_nhwc(const float* x, float* xp, int b, int hi, int wi,
                     int c, hipStream_t stream);
void launch_conv3x3_fwd_fast(const float* xp, const float* wp, float* y,
                             int b, int hi, int wi, int ho, int wo, int k,
                             int n, int stride, hipStream_t stream);
void launch_conv3x3_fwd_nopad(const float* x, const float* wp, float* y,
                              const float* zpage, int b, int hi, int wi,
                              int ho, int wo, int k, int n, int stride,
                              hipStream_t stream);
void launch_conv3x3_wgrad(const float* dy, const float* xp, float* dw9,
                          float* dw, int b, int hi, int wi, int ho, int wo,
                          int k, int n, int stride, hipStream_t stream);
void launch_lars_momentum_step(float* p, const float* g, float* m,
                               float* norm_acc, float* alr,
                               const int64_t* seg_off,
                               const int64_t* seg_len, const float* seg_wd,
                               const int* seg_adapt, const int* chunk_seg,
                               const int64_t* chunk_base, int nseg,
                               int nchunks, int64_t chunk, float trust,
                               float eps, float lr, const float* lr_dev,
                               float momentum,
                               int m_init, hipStream_t stream);

namespace {

#define CHECK_IN(t)                                                        \
  TORCH_CHECK((t).is_cuda(), #t " must be on GPU");                        \
  TORCH_CHECK((t).is_contiguous(), #t " must be contiguous");              \
  TORCH_CHECK((t).scalar_type() == at::kFloat, #t " must be fp32")

void flat_ema_update(torch::Tensor mean, torch::Tensor x, double decay) {
  CHECK_IN(mean);
  CHECK_IN(x);
  TORCH_CHECK(mean.numel() == x.numel(), "size mismatch");
  auto stream = at::hip::getCurrentHIPStream();
  launch_flat_ema_update(mean.data_ptr<float>(), x.data_ptr<float>(),
                         static_cast<float>(decay), nullptr, mean.numel(),
                         stream);
}

// decay read from a 1-float device scalar: hipGraph-capturable (the replay
// wrapper rewrites the scalar between replays, outside the graph)
void flat_ema_update_dev(torch::Tensor mean, torch::Tensor x,
                         torch::Tensor decay_dev) {
  CHECK_IN(mean);
  CHECK_IN(x);
  CHECK_IN(decay_dev);
  TORCH_CHECK(mean.numel() == x.numel(), "size mismatch");
  TORCH_CHECK(decay_dev.numel() == 1, "decay_dev must be a 1-elem tensor");
  auto stream = at::hip::getCurrentHIPStream();
  launch_flat_ema_update(mean.data_ptr<float>(), x.data_ptr<float>(), 0.f,
                         decay_dev.data_ptr<float>(), mean.numel(), stream);
}

std::tuple<torch::Tensor, torch::Tensor> byol_loss_forward(
    torch::Tensor p1, torch::Tensor p2, torch::Tensor z1, torch::Tensor z2) {
  CHECK_IN(p1); CHECK_IN(p2); CHECK_IN(z1); CHECK_IN(z2);
  TORCH_CHECK(p1.dim() == 2, "expected 2-D [B, D]");
  TORCH_CHECK(p1.sizes() == p2.sizes() && p1.sizes() == z1.sizes() &&
              p1.sizes() == z2.sizes(), "shape mismatch");
  const int64_t n = p1.numel();
  const int64_t batch = p1.size(0);
  auto stats = torch::zeros({6}, p1.options());
  auto loss = torch::empty({}, p1.options());
  auto stream = at::hip::getCurrentHIPStream();
  launch_byol_loss_forward(p1.data_ptr<float>(), p2.data_ptr<float>(),
                           z1.data_ptr<float>(), z2.data_ptr<float>(),
                           stats.data_ptr<float>(), loss.data_ptr<float>(), n,
                           batch, stream);
  return {loss, stats};
}

std::tuple<torch::Tensor, torch::Tensor> byol_loss_backward(
    torch::Tensor p1, torch::Tensor p2, torch::Tensor z1, torch::Tensor z2,
    torch::Tensor stats, torch::Tensor grad_out) {
  CHECK_IN(p1); CHECK_IN(p2); CHECK_IN(z1); CHECK_IN(z2); CHECK_IN(stats);
  TORCH_CHECK(grad_out.is_cuda() && grad_out.scalar_type() == at::kFloat,
              "grad_out must be fp32 on GPU");
  const int64_t n = p1.numel();
  const int64_t batch = p1.size(0);
  auto g1 = torch::empty_like(p1);
  auto g2 = torch::empty_like(p2);
  auto stream = at::hip::getCurrentHIPStream();
  launch_byol_loss_backward(
      p1.data_ptr<float>(), p2.data_ptr<float>(), z1.data_ptr<float>(),
      z2.data_ptr<float>(), stats.data_ptr<float>(),
      grad_out.contiguous().data_ptr<float>(), g1.data_ptr<float>(),
      g2.data_ptr<float>(), n, batch, stream);
  return {g1, g2};
}

void conv3x3_fwd_nopad(torch::Tensor x, torch::Tensor wp, torch::Tensor y,
                       torch::Tensor zpage, int64_t b, int64_t hi,
                       int64_t wi, int64_t ho, int64_t wo, int64_t k,
                       int64_t n, int64_t stride) {
  CHECK_IN(x); CHECK_IN(wp); CHECK_IN(y); CHECK_IN(zpage);
  TORCH_CHECK(zpage.numel() >= 4, "zpage must hold >= 16 bytes");
  auto stream = at::hip::getCurrentHIPStream();
  launch_conv3x3_fwd_nopad(x.data_ptr<float>(), wp.data_ptr<float>(),
                           y.data_ptr<float>(), zpage.data_ptr<float>(),
                           (int)b, (int)hi, (int)wi, (int)ho, (int)wo,
                           (int)k, (int)n, (int)stride, stream);
}

void conv3x3_wgrad(torch::Tensor dy, torch::Tensor xp, torch::Tensor dw9,
                   torch::Tensor dw, int64_t b, int64_t hi, int64_t wi,
                   int64_t ho, int64_t wo, int64_t k, int64_t n,
                   int64_t stride) {
  CHECK_IN(dy); CHECK_IN(xp); CHECK_IN(dw9); CHECK_IN(dw);
  TORCH_CHECK(dw9.numel() == 9 * n * k, "dw9 must be [9][N][K]");
  TORCH_CHECK(dw.numel() == n * k * 9, "dw must be [N][K][3][3]");
  auto stream = at::hip::getCurrentHIPStream();
  launch_conv3x3_wgrad(dy.data_ptr<float>(), xp.data_ptr<float>(),
                       dw9.data_ptr<float>(), dw.data_ptr<float>(), (int)b,
                       (int)hi, (int)wi, (int)ho, (int)wo, (int)k, (int)n,
                       (int)stride, stream);
}

void lars_momentum_step(torch::Tensor p, torch::Tensor g, torch::Tensor m,
                        torch::Tensor norm_acc, torch::Tensor alr,
                        torch::Tensor seg_off, torch::Tensor seg_len,
                        torch::Tensor seg_wd, torch::Tensor seg_adapt,
                        torch::Tensor chunk_seg, torch::Tensor chunk_base,
                        double trust, double eps, double lr, double momentum,
                        int64_t m_init,
                        c10::optional<torch::Tensor> lr_dev = c10::nullopt) {
  CHECK_IN(p); CHECK_IN(g); CHECK_IN(m); CHECK_IN(norm_acc); CHECK_IN(alr);
  TORCH_CHECK(seg_off.scalar_type() == at::kLong &&
              seg_len.scalar_type() == at::kLong &&
              chunk_base.scalar_type() == at::kLong, "bad index dtypes");
  TORCH_CHECK(seg_adapt.scalar_type() == at::kInt &&
              chunk_seg.scalar_type() == at::kInt, "bad flag dtypes");
  const int nseg = seg_off.numel();
  const int nchunks = chunk_seg.numel();
  auto stream = at::hip::getCurrentHIPStream();
  launch_lars_momentum_step(
      p.data_ptr<float>(), g.data_ptr<float>(), m.data_ptr<float>(),
      norm_acc.data_ptr<float>(), alr.data_ptr<float>(),
      seg_off.data_ptr<int64_t>(), seg_len.data_ptr<int64_t>(),
      seg_wd.data_ptr<float>(), seg_adapt.data_ptr<int>(),
      chunk_seg.data_ptr<int>(), chunk_base.data_ptr<int64_t>(), nseg,
      nchunks, 65536, static_cast<float>(trust), static_cast<float>(eps),
      static_cast<float>(lr),
      lr_dev.has_value() ? lr_dev->data_ptr<float>() : nullptr,
      static_cast<float>(momentum),
      static_cast<int>(m_init), stream);
}

// --- fused BN(+residual)(+ReLU), NHWC/2-D rows-x-channels fp32 -------------
// All tensors are raw [M, C]-layout fp32 views prepared by the Python
// wrapper (byol_amd/ops/bn.py), which owns layout checks and SyncBN comm.

void bn_stats(torch::Tensor x, torch::Tensor acc, int64_t m, int64_t c,
              int64_t nslots) {
  CHECK_IN(x); CHECK_IN(acc);
  TORCH_CHECK(acc.numel() == nslots * 2 * c, "acc must be [nslots][2C]");
  auto stream = at::hip::getCurrentHIPStream();
  launch_bn_stats(x.data_ptr<float>(), acc.data_ptr<float>(), m, (int)c,
                  (int)(nslots - 1), stream);
}

void bn_stats_v2(torch::Tensor x, torch::Tensor acc, int64_t m, int64_t c,
                 int64_t nslots, int64_t grid) {
  CHECK_IN(x); CHECK_IN(acc);
  TORCH_CHECK(acc.numel() == nslots * 2 * c, "acc must be [nslots][2C]");
  auto stream = at::hip::getCurrentHIPStream();
  launch_bn_stats_v2(x.data_ptr<float>(), acc.data_ptr<float>(), m, (int)c,
                     (int)(nslots - 1), (int)grid, stream);
}

void bn_bwd_reduce_v2(torch::Tensor dy, torch::Tensor y, torch::Tensor x,
                      torch::Tensor mean, torch::Tensor invstd,
                      torch::Tensor red, int64_t m, int64_t c,
                      int64_t relu, int64_t nslots, int64_t grid) {
  CHECK_IN(dy); CHECK_IN(y); CHECK_IN(x); CHECK_IN(red);
  auto stream = at::hip::getCurrentHIPStream();
  launch_bn_bwd_reduce_v2(dy.data_ptr<float>(), y.data_ptr<float>(),
                          x.data_ptr<float>(), mean.data_ptr<float>(),
                          invstd.data_ptr<float>(), red.data_ptr<float>(),
                          m, (int)c, (int)relu, (int)(nslots - 1),
                          (int)grid, stream);
}

void bn_apply_v2(torch::Tensor x, c10::optional<torch::Tensor> residual,
                 torch::Tensor mean, torch::Tensor invstd,
                 torch::Tensor weight, torch::Tensor bias, torch::Tensor y,
                 int64_t m, int64_t c, int64_t relu, int64_t grid) {
  CHECK_IN(x); CHECK_IN(y);
  auto stream = at::hip::getCurrentHIPStream();
  launch_bn_apply_v2(
      x.data_ptr<float>(),
      residual.has_value() ? residual->data_ptr<float>() : nullptr,
      mean.data_ptr<float>(), invstd.data_ptr<float>(),
      weight.data_ptr<float>(), bias.data_ptr<float>(),
      y.data_ptr<float>(), m, (int)c, (int)relu, (int)grid, stream);
}

void bn_bwd_apply_v2(torch::Tensor dy, torch::Tensor y, torch::Tensor x,
                     torch::Tensor mean, torch::Tensor invstd,
                     torch::Tensor weight, torch::Tensor red,
                     torch::Tensor dx, c10::optional<torch::Tensor> dres,
                     double inv_count, int64_t m, int64_t c, int64_t relu,
                     int64_t grid) {
  CHECK_IN(dy); CHECK_IN(dx);
  auto stream = at::hip::getCurrentHIPStream();
  launch_bn_bwd_apply_v2(
      dy.data_ptr<float>(), y.data_ptr<float>(), x.data_ptr<float>(),
      mean.data_ptr<float>(), invstd.data_ptr<float>(),
      weight.data_ptr<float>(), red.data_ptr<float>(),
      dx.data_ptr<float>(),
      dres.has_value() ? dres->data_ptr<float>() : nullptr,
      (float)inv_count, m, (int)c, (int)relu, (int)grid, stream);
}

void bn_reduce_slots(torch::Tensor in, torch::Tensor out, int64_t nslots) {
  CHECK_IN(in); CHECK_IN(out);
  auto stream = at::hip::getCurrentHIPStream();
  launch_bn_reduce_slots(in.data_ptr<float>(), out.data_ptr<float>(),
                         (int)out.numel(), (int)nslots, stream);
}

void bn_finalize(torch::Tensor acc, torch::Tensor mean, torch::Tensor invstd,
                 torch::Tensor running_mean, torch::Tensor running_var,
                 double count, double eps, double momentum, int64_t c,
                 int64_t update_running) {
  CHECK_IN(acc); CHECK_IN(mean); CHECK_IN(invstd);
  auto stream = at::hip::getCurrentHIPStream();
  launch_bn_finalize(acc.data_ptr<float>(), mean.data_ptr<float>(),
                     invstd.data_ptr<float>(),
                     update_running ? running_mean.data_ptr<float>() : nullptr,
                     update_running ? running_var.data_ptr<float>() : nullptr,
                     (float)count, (float)eps, (float)momentum, (int)c,
                     (int)update_running, stream);
}

void bn_apply(torch::Tensor x, c10::optional<torch::Tensor> residual,
              torch::Tensor mean, torch::Tensor invstd, torch::Tensor weight,
              torch::Tensor bias, torch::Tensor y, int64_t m, int64_t c,
              int64_t relu) {
  CHECK_IN(x); CHECK_IN(mean); CHECK_IN(invstd); CHECK_IN(weight);
  CHECK_IN(bias); CHECK_IN(y);
  auto stream = at::hip::getCurrentHIPStream();
  launch_bn_apply(x.data_ptr<float>(),
                  residual ? residual->data_ptr<float>() : nullptr,
                  mean.data_ptr<float>(), invstd.data_ptr<float>(),
                  weight.data_ptr<float>(), bias.data_ptr<float>(),
                  y.data_ptr<float>(), m, (int)c, (int)relu, stream);
}

void bn_bwd_reduce(torch::Tensor dy, torch::Tensor y, torch::Tensor x,
                   torch::Tensor mean, torch::Tensor invstd,
                   torch::Tensor red, int64_t m, int64_t c, int64_t relu,
                   int64_t nslots) {
  CHECK_IN(dy); CHECK_IN(y); CHECK_IN(x); CHECK_IN(red);
  TORCH_CHECK(red.numel() == nslots * 2 * c, "red must be [nslots][2C]");
  auto stream = at::hip::getCurrentHIPStream();
  launch_bn_bwd_reduce(dy.data_ptr<float>(), y.data_ptr<float>(),
                       x.data_ptr<float>(), mean.data_ptr<float>(),
                       invstd.data_ptr<float>(), red.data_ptr<float>(), m,
                       (int)c, (int)relu, (int)(nslots - 1), stream);
}

void bn_bwd_apply(torch::Tensor dy, torch::Tensor y, torch::Tensor x,
                  torch::Tensor mean, torch::Tensor invstd,
                  torch::Tensor weight, torch::Tensor red, torch::Tensor dx,
                  c10::optional<torch::Tensor> dresidual, double inv_count,
                  int64_t m, int64_t c, int64_t relu) {
  CHECK_IN(dy); CHECK_IN(y); CHECK_IN(x); CHECK_IN(dx);
  auto stream = at::hip::getCurrentHIPStream();
  launch_bn_bwd_apply(dy.data_ptr<float>(), y.data_ptr<float>(),
                      x.data_ptr<float>(), mean.data_ptr<float>(),
                      invstd.data_ptr<float>(), weight.data_ptr<float>(),
                      red.data_ptr<float>(), dx.data_ptr<float>(),
                      dresidual ? dresidual->data_ptr<float>() : nullptr,
                      (float)inv_count, m, (int)c, (int)relu, stream);
}

void aug_sample(torch::Tensor src, torch::Tensor dst, torch::Tensor gray_sum,
                torch::Tensor crop, int64_t hs, int64_t ws, int64_t s,
                int64_t use_v2) {
  CHECK_IN(src); CHECK_IN(dst); CHECK_IN(gray_sum); CHECK_IN(crop);
  const int b = gray_sum.numel();
  auto stream = at::hip::getCurrentHIPStream();
  launch_aug_sample(src.data_ptr<float>(), dst.data_ptr<float>(),
                    gray_sum.data_ptr<float>(), crop.data_ptr<float>(), b,
                    (int)hs, (int)ws, (int)s, (int)use_v2, stream);
}

void aug_color(torch::Tensor img, torch::Tensor gray_sum,
               torch::Tensor cparam, int64_t s) {
  CHECK_IN(img); CHECK_IN(gray_sum); CHECK_IN(cparam);
  const int b = gray_sum.numel();
  auto stream = at::hip::getCurrentHIPStream();
  launch_aug_color(img.data_ptr<float>(), gray_sum.data_ptr<float>(),
                   cparam.data_ptr<float>(), b, (int)s, stream);
}

void aug_blur(torch::Tensor img, torch::Tensor tmp, torch::Tensor out,
              torch::Tensor sigma, torch::Tensor wts, int64_t s,
              int64_t ksize) {
  CHECK_IN(img); CHECK_IN(tmp); CHECK_IN(out); CHECK_IN(sigma);
  CHECK_IN(wts);
  const int b = (int)sigma.numel();
  TORCH_CHECK(img.numel() == (int64_t)b * s * s * 3, "img shape mismatch");
  TORCH_CHECK(wts.numel() >= (int64_t)b * ksize, "wts too small");
  auto stream = at::hip::getCurrentHIPStream();
  launch_aug_blur(img.data_ptr<float>(), tmp.data_ptr<float>(),
                  out.data_ptr<float>(), sigma.data_ptr<float>(),
                  wts.data_ptr<float>(), b, (int)s, (int)ksize, stream);
}

void ce_topk_fwd(torch::Tensor logits, torch::Tensor labels,
                 torch::Tensor out, torch::Tensor row_stats) {
  CHECK_IN(logits); CHECK_IN(out); CHECK_IN(row_stats);
  TORCH_CHECK(labels.scalar_type() == at::kLong, "labels must be int64");
  auto stream = at::hip::getCurrentHIPStream();
  launch_ce_topk_fwd(logits.data_ptr<float>(), labels.data_ptr<int64_t>(),
                     out.data_ptr<float>(), row_stats.data_ptr<float>(),
                     (int)logits.size(0), (int)logits.size(1), stream);
}

void ce_bwd(torch::Tensor logits, torch::Tensor labels,
            torch::Tensor row_stats, torch::Tensor grad_out,
            torch::Tensor dlogits) {
  CHECK_IN(logits); CHECK_IN(row_stats); CHECK_IN(dlogits);
  auto stream = at::hip::getCurrentHIPStream();
  launch_ce_bwd(logits.data_ptr<float>(), labels.data_ptr<int64_t>(),
                row_stats.data_ptr<float>(), grad_out.data_ptr<float>(),
                dlogits.data_ptr<float>(), (int)logits.size(0),
                (int)logits.size(1), stream);
}

void conv1x1_fwd(torch::Tensor x, torch::Tensor w,
                 c10::optional<torch::Tensor> wt, torch::Tensor y,
                 int64_t m, int64_t k, int64_t n) {
  CHECK_IN(x); CHECK_IN(w); CHECK_IN(y);
  auto stream = at::hip::getCurrentHIPStream();
  launch_conv1x1_fwd(x.data_ptr<float>(), w.data_ptr<float>(),
                     wt ? wt->data_ptr<float>() : nullptr,
                     y.data_ptr<float>(), m, (int)k, (int)n, stream);
}

void conv1x1_dgrad(torch::Tensor dy, torch::Tensor w, torch::Tensor dx,
                   int64_t m, int64_t n, int64_t k) {
  CHECK_IN(dy); CHECK_IN(w); CHECK_IN(dx);
  auto stream = at::hip::getCurrentHIPStream();
  launch_conv1x1_dgrad(dy.data_ptr<float>(), w.data_ptr<float>(),
                       dx.data_ptr<float>(), m, (int)n, (int)k, stream);
}

void conv1x1_wgrad(torch::Tensor dy, torch::Tensor x, torch::Tensor dw,
                   int64_t m, int64_t n, int64_t k) {
  CHECK_IN(dy); CHECK_IN(x); CHECK_IN(dw);
  auto stream = at::hip::getCurrentHIPStream();
  launch_conv1x1_wgrad(dy.data_ptr<float>(), x.data_ptr<float>(),
                       dw.data_ptr<float>(), m, (int)n, (int)k, stream);
}

int64_t wgrad_nchunks(int64_t m, int64_t n, int64_t k) {
  return conv1x1_wgrad_nchunks(m, (int)n, (int)k);
}

void conv1x1_wgrad_v2(torch::Tensor dy, torch::Tensor x,
                      torch::Tensor partial, torch::Tensor dw, int64_t m,
                      int64_t n, int64_t k) {
  CHECK_IN(dy); CHECK_IN(x); CHECK_IN(partial); CHECK_IN(dw);
  TORCH_CHECK(partial.numel() >= wgrad_nchunks(m, n, k) * n * k,
              "partial buffer too small");
  auto stream = at::hip::getCurrentHIPStream();
  launch_conv1x1_wgrad_partial(dy.data_ptr<float>(), x.data_ptr<float>(),
                               partial.data_ptr<float>(),
                               dw.data_ptr<float>(), m, (int)n, (int)k,
                               stream);
}

void pad_nhwc(torch::Tensor x, torch::Tensor xp, int64_t b, int64_t hi,
              int64_t wi, int64_t c) {
  CHECK_IN(x); CHECK_IN(xp);
  auto stream = at::hip::getCurrentHIPStream();
  launch_pad_nhwc(x.data_ptr<float>(), xp.data_ptr<float>(), (int)b,
                  (int)hi, (int)wi, (int)c, stream);
}

void conv3x3_fwd_fast(torch::Tensor xp, torch::Tensor wp, torch::Tensor y,
                      int64_t b, int64_t hi, int64_t wi, int64_t ho,
                      int64_t wo, int64_t k, int64_t n, int64_t stride) {
  CHECK_IN(xp); CHECK_IN(wp); CHECK_IN(y);
  auto stream = at::hip::getCurrentHIPStream();
  launch_conv3x3_fwd_fast(xp.data_ptr<float>(), wp.data_ptr<float>(),
                          y.data_ptr<float>(), (int)b, (int)hi, (int)wi,
                          (int)ho, (int)wo, (int)k, (int)n, (int)stride,
                          stream);
}

void conv3x3_fwd(torch::Tensor x, torch::Tensor wp, torch::Tensor y,
                 int64_t b, int64_t hi, int64_t wi, int64_t ho, int64_t wo,
                 int64_t k, int64_t n, int64_t stride) {
  CHECK_IN(x); CHECK_IN(wp); CHECK_IN(y);
  auto stream = at::hip::getCurrentHIPStream();
  launch_conv3x3_fwd(x.data_ptr<float>(), wp.data_ptr<float>(),
                     y.data_ptr<float>(), (int)b, (int)hi, (int)wi, (int)ho,
                     (int)wo, (int)k, (int)n, (int)stride, stream);
}


// --- bf16-I/O fused BN (round-2 candidate; stats/params stay fp32) --------

#define CHECK_BF(t)                                                        \
  TORCH_CHECK((t).is_cuda(), #t " must be on GPU");                        \
  TORCH_CHECK((t).is_contiguous(), #t " must be contiguous");              \
  TORCH_CHECK((t).scalar_type() == at::kBFloat16, #t " must be bf16")

static const uint16_t* bfp(const torch::Tensor& t) {
  return reinterpret_cast<const uint16_t*>(t.data_ptr<at::BFloat16>());
}
static uint16_t* bfp_mut(torch::Tensor& t) {
  return reinterpret_cast<uint16_t*>(t.data_ptr<at::BFloat16>());
}

void bn_stats_bf16(torch::Tensor x, torch::Tensor acc, int64_t m, int64_t c,
                   int64_t nslots) {
  CHECK_BF(x); CHECK_IN(acc);
  auto stream = at::hip::getCurrentHIPStream();
  launch_bn_stats_bf16(bfp(x), acc.data_ptr<float>(), m, (int)c,
                       (int)(nslots - 1), stream);
}

void bn_apply_bf16(torch::Tensor x, c10::optional<torch::Tensor> residual,
                   torch::Tensor mean, torch::Tensor invstd,
                   torch::Tensor weight, torch::Tensor bias, torch::Tensor y,
                   int64_t m, int64_t c, int64_t relu) {
  CHECK_BF(x); CHECK_BF(y);
  CHECK_IN(mean); CHECK_IN(invstd); CHECK_IN(weight); CHECK_IN(bias);
  auto stream = at::hip::getCurrentHIPStream();
  launch_bn_apply_bf16(bfp(x), residual ? bfp(*residual) : nullptr,
                       mean.data_ptr<float>(), invstd.data_ptr<float>(),
                       weight.data_ptr<float>(), bias.data_ptr<float>(),
                       bfp_mut(y), m, (int)c, (int)relu, stream);
}

void bn_bwd_reduce_bf16(torch::Tensor dy, torch::Tensor y, torch::Tensor x,
                        torch::Tensor mean, torch::Tensor invstd,
                        torch::Tensor red, int64_t m, int64_t c,
                        int64_t relu, int64_t nslots) {
  CHECK_BF(dy); CHECK_BF(y); CHECK_BF(x); CHECK_IN(red);
  auto stream = at::hip::getCurrentHIPStream();
  launch_bn_bwd_reduce_bf16(bfp(dy), bfp(y), bfp(x),
                            mean.data_ptr<float>(),
                            invstd.data_ptr<float>(), red.data_ptr<float>(),
                            m, (int)c, (int)relu, (int)(nslots - 1), stream);
}

void bn_bwd_apply_bf16(torch::Tensor dy, torch::Tensor y, torch::Tensor x,
                       torch::Tensor mean, torch::Tensor invstd,
                       torch::Tensor weight, torch::Tensor red,
                       torch::Tensor dx, c10::optional<torch::Tensor> dres,
                       double inv_count, int64_t m, int64_t c,
                       int64_t relu) {
  CHECK_BF(dy); CHECK_BF(y); CHECK_BF(x); CHECK_BF(dx);
  auto stream = at::hip::getCurrentHIPStream();
  torch::Tensor drt;
  uint16_t* drp = nullptr;
  if (dres) { drt = *dres; drp = bfp_mut(drt); }
  launch_bn_bwd_apply_bf16(bfp(dy), bfp(y), bfp(x), mean.data_ptr<float>(),
                           invstd.data_ptr<float>(),
                           weight.data_ptr<float>(), red.data_ptr<float>(),
                           bfp_mut(dx), drp, (float)inv_count, m, (int)c,
                           (int)relu, stream);
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, mod) {
  mod.def("bn_stats_bf16", &bn_stats_bf16);
  mod.def("bn_apply_bf16", &bn_apply_bf16);
  mod.def("bn_bwd_reduce_bf16", &bn_bwd_reduce_bf16);
  mod.def("bn_bwd_apply_bf16", &bn_bwd_apply_bf16);
  mod.def("conv3x3_fwd", &conv3x3_fwd);
  mod.def("conv3x3_fwd_fast", &conv3x3_fwd_fast);
  mod.def("conv3x3_wgrad", &conv3x3_wgrad);
  mod.def("conv3x3_fwd_nopad", &conv3x3_fwd_nopad);
  mod.def("pad_nhwc", &pad_nhwc);
  mod.def("conv1x1_fwd", &conv1x1_fwd);
  mod.def("conv1x1_dgrad", &conv1x1_dgrad);
  mod.def("conv1x1_wgrad", &conv1x1_wgrad);
  mod.def("conv1x1_wgrad_v2", &conv1x1_wgrad_v2);
  mod.def("wgrad_nchunks", &wgrad_nchunks);
  mod.def("ce_topk_fwd", &ce_topk_fwd);
  mod.def("ce_bwd", &ce_bwd);
  mod.def("aug_sample", &aug_sample);
  mod.def("aug_color", &aug_color);
  mod.def("aug_blur", &aug_blur);
  mod.def("bn_stats", &bn_stats);
  mod.def("bn_stats_v2", &bn_stats_v2);
  mod.def("bn_bwd_reduce_v2", &bn_bwd_reduce_v2);
  mod.def("bn_apply_v2", &bn_apply_v2);
  mod.def("bn_bwd_apply_v2", &bn_bwd_apply_v2);
  mod.def("bn_reduce_slots", &bn_reduce_slots);
  mod.def("bn_finalize", &bn_finalize);
  mod.def("bn_apply", &bn_apply);
  mod.def("bn_bwd_reduce", &bn_bwd_reduce);
  mod.def("bn_bwd_apply", &bn_bwd_apply);
  mod.def("flat_ema_update", &flat_ema_update,
          "fused flat-parameter EMA update (gfx950)");
  mod.def("flat_ema_update_dev", &flat_ema_update_dev,
          "EMA update with device-scalar decay (hipGraph-capturable)");
  mod.def("byol_loss_forward", &byol_loss_forward,
          "fused BYOL loss forward (gfx950)");
  mod.def("byol_loss_backward", &byol_loss_backward,
          "fused BYOL loss backward (gfx950)");
  mod.def("lars_momentum_step", &lars_momentum_step,
          "fused multi-tensor LARS+momentum step (gfx950)",
          pybind11::arg("p"), pybind11::arg("g"), pybind11::arg("m"),
          pybind11::arg("norm_acc"), pybind11::arg("alr"),
          pybind11::arg("seg_off"), pybind11::arg("seg_len"),
          pybind11::arg("seg_wd"), pybind11::arg("seg_adapt"),
          pybind11::arg("chunk_seg"), pybind11::arg("chunk_base"),
          pybind11::arg("trust"), pybind11::arg("eps"), pybind11::arg("lr"),
          pybind11::arg("momentum"), pybind11::arg("m_init"),
          pybind11::arg("lr_dev") = pybind11::none());
}
