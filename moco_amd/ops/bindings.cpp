// Python <-> HIP bindings for moco_amd._C.
// This is the only torch-dependent TU; the kernels live in hip/*.hip behind
// the C ABI of hip/launchers.h.
#include <torch/extension.h>

#include <ATen/cuda/CUDAContext.h>
#include <c10/cuda/CUDAGuard.h>

#include "hip/launchers.h"

namespace {

void* cur_stream() {
  return (void*)at::cuda::getCurrentCUDAStream().stream();
}

#define CHECK_LAUNCH(expr)                                            \
  do {                                                                \
    int rc_ = (expr);                                                 \
    TORCH_CHECK(rc_ == 0, #expr " failed with hipError_t=", rc_);     \
  } while (0)

#define CHECK_CUDA_CONTIG(t)                                          \
  TORCH_CHECK((t).is_cuda() && (t).is_contiguous(), #t,               \
              " must be a contiguous GPU tensor")

bool is_bf16(const torch::Tensor& t) {
  return t.scalar_type() == torch::kBFloat16;
}

// ---------------------------------------------------------------------------
std::vector<torch::Tensor> l2norm_rows_fwd(torch::Tensor x) {
  CHECK_CUDA_CONTIG(x);
  TORCH_CHECK(x.dim() == 2, "x must be [N, C]");
  TORCH_CHECK(is_bf16(x) || x.scalar_type() == torch::kFloat32);
  at::cuda::CUDAGuard guard(x.device());
  auto y = torch::empty_like(x);
  auto inv_norm = torch::empty({x.size(0)},
                               x.options().dtype(torch::kFloat32));
  CHECK_LAUNCH(launch_l2norm_fwd(cur_stream(), x.data_ptr(), y.data_ptr(),
                                 inv_norm.data_ptr<float>(), x.size(0),
                                 x.size(1), is_bf16(x)));
  return {y, inv_norm};
}

torch::Tensor l2norm_rows_bwd(torch::Tensor g, torch::Tensor y,
                              torch::Tensor inv_norm) {
  CHECK_CUDA_CONTIG(g);
  CHECK_CUDA_CONTIG(y);
  TORCH_CHECK(g.sizes() == y.sizes() && g.scalar_type() == y.scalar_type());
  at::cuda::CUDAGuard guard(g.device());
  auto gx = torch::empty_like(g);
  CHECK_LAUNCH(launch_l2norm_bwd(cur_stream(), g.data_ptr(), y.data_ptr(),
                                 inv_norm.data_ptr<float>(), gx.data_ptr(),
                                 g.size(0), g.size(1), is_bf16(g)));
  return gx;
}

// ---------------------------------------------------------------------------
torch::Tensor contrastive_fwd(torch::Tensor q, torch::Tensor k,
                              torch::Tensor queue_kc, double invT) {
  CHECK_CUDA_CONTIG(q);
  CHECK_CUDA_CONTIG(k);
  CHECK_CUDA_CONTIG(queue_kc);
  TORCH_CHECK(is_bf16(q) && is_bf16(k) && is_bf16(queue_kc),
              "contrastive_fwd expects bf16 q/k/queue_kc");
  const int64_t N = q.size(0), C = q.size(1), K = queue_kc.size(0);
  TORCH_CHECK(k.size(0) == N && k.size(1) == C && queue_kc.size(1) == C);
  at::cuda::CUDAGuard guard(q.device());
  auto out = torch::empty({N, 1 + K}, q.options().dtype(torch::kFloat32));
  CHECK_LAUNCH(launch_contrastive_fwd(cur_stream(), q.data_ptr(),
                                      k.data_ptr(), queue_kc.data_ptr(),
                                      out.data_ptr<float>(), N, C, K,
                                      (float)invT));
  return out;
}

torch::Tensor contrastive_bwd(torch::Tensor g, torch::Tensor k,
                              torch::Tensor queue_ck, double invT) {
  CHECK_CUDA_CONTIG(g);
  CHECK_CUDA_CONTIG(k);
  CHECK_CUDA_CONTIG(queue_ck);
  TORCH_CHECK(g.scalar_type() == torch::kFloat32 &&
              queue_ck.scalar_type() == torch::kFloat32 && is_bf16(k));
  const int64_t N = g.size(0), C = queue_ck.size(0), K = queue_ck.size(1);
  TORCH_CHECK(g.size(1) == 1 + K && k.size(0) == N && k.size(1) == C);
  at::cuda::CUDAGuard guard(g.device());
  auto dq = torch::zeros({N, C}, g.options());
  CHECK_LAUNCH(launch_contrastive_bwd(cur_stream(), g.data_ptr<float>(),
                                      k.data_ptr(),
                                      queue_ck.data_ptr<float>(),
                                      dq.data_ptr<float>(), N, C, K,
                                      (float)invT));
  return dq;
}

// ---------------------------------------------------------------------------
std::vector<torch::Tensor> softmax_ce0_fwd(torch::Tensor logits) {
  CHECK_CUDA_CONTIG(logits);
  TORCH_CHECK(logits.scalar_type() == torch::kFloat32 && logits.dim() == 2);
  at::cuda::CUDAGuard guard(logits.device());
  const int64_t N = logits.size(0), M = logits.size(1);
  auto opts = logits.options();
  auto loss_rows = torch::empty({N}, opts);
  auto row_max = torch::empty({N}, opts);
  auto row_lse = torch::empty({N}, opts);
  CHECK_LAUNCH(launch_softmax_ce0_fwd(
      cur_stream(), logits.data_ptr<float>(), loss_rows.data_ptr<float>(),
      row_max.data_ptr<float>(), row_lse.data_ptr<float>(), N, M));
  return {loss_rows, row_max, row_lse};
}

// gscale (optional fp32 scalar tensor): extra scale read on device — the
// upstream grad under hipGraph capture, where a host .item() would sync
torch::Tensor softmax_ce0_bwd(torch::Tensor logits, torch::Tensor row_max,
                              torch::Tensor row_lse, double scale,
                              c10::optional<torch::Tensor> gscale) {
  CHECK_CUDA_CONTIG(logits);
  at::cuda::CUDAGuard guard(logits.device());
  auto dlogits = torch::empty_like(logits);
  const float* gs = nullptr;
  if (gscale.has_value()) {
    TORCH_CHECK(gscale->is_cuda() &&
                gscale->scalar_type() == torch::kFloat32 &&
                gscale->numel() == 1);
    gs = gscale->data_ptr<float>();
  }
  CHECK_LAUNCH(launch_softmax_ce0_bwd(
      cur_stream(), logits.data_ptr<float>(), row_max.data_ptr<float>(),
      row_lse.data_ptr<float>(), dlogits.data_ptr<float>(), (float)scale,
      gs, logits.size(0), logits.size(1)));
  return dlogits;
}

// ---------------------------------------------------------------------------
// graph-replay-safe enqueue: ring position read/advanced from the queue_ptr
// buffer on device (a host ptr would be frozen into a captured graph)
void queue_enqueue_dev(torch::Tensor queue,
                       c10::optional<torch::Tensor> queue_kc,
                       torch::Tensor keys, torch::Tensor queue_ptr) {
  CHECK_CUDA_CONTIG(queue);
  CHECK_CUDA_CONTIG(keys);
  TORCH_CHECK(queue.scalar_type() == torch::kFloat32);
  TORCH_CHECK(queue_ptr.is_cuda() &&
              queue_ptr.scalar_type() == torch::kInt64);
  const int64_t C = queue.size(0), K = queue.size(1), B = keys.size(0);
  TORCH_CHECK(keys.size(1) == C && K % B == 0);
  void* kc = nullptr;
  if (queue_kc.has_value()) {
    CHECK_CUDA_CONTIG(*queue_kc);
    TORCH_CHECK(is_bf16(*queue_kc) && queue_kc->size(0) == K &&
                queue_kc->size(1) == C);
    kc = queue_kc->data_ptr();
  }
  at::cuda::CUDAGuard guard(queue.device());
  CHECK_LAUNCH(launch_queue_enqueue_dev(
      cur_stream(), queue.data_ptr<float>(), kc, keys.data_ptr(),
      is_bf16(keys), C, K, B, queue_ptr.data_ptr<int64_t>()));
}

void queue_enqueue(torch::Tensor queue, c10::optional<torch::Tensor> queue_kc,
                   torch::Tensor keys, int64_t ptr) {
  CHECK_CUDA_CONTIG(queue);
  CHECK_CUDA_CONTIG(keys);
  TORCH_CHECK(queue.scalar_type() == torch::kFloat32);
  const int64_t C = queue.size(0), K = queue.size(1), B = keys.size(0);
  TORCH_CHECK(keys.size(1) == C && ptr + B <= K);
  void* kc = nullptr;
  if (queue_kc.has_value()) {
    CHECK_CUDA_CONTIG(*queue_kc);
    TORCH_CHECK(is_bf16(*queue_kc) && queue_kc->size(0) == K &&
                queue_kc->size(1) == C);
    kc = queue_kc->data_ptr();
  }
  at::cuda::CUDAGuard guard(queue.device());
  CHECK_LAUNCH(launch_queue_enqueue(cur_stream(),
                                    queue.data_ptr<float>(), kc,
                                    keys.data_ptr(), is_bf16(keys), C, K, B,
                                    ptr));
}

// ---------------------------------------------------------------------------
void multi_tensor_ema(torch::Tensor pd, torch::Tensor ps, torch::Tensor sizes,
                      double m) {
  CHECK_CUDA_CONTIG(pd);
  CHECK_CUDA_CONTIG(ps);
  CHECK_CUDA_CONTIG(sizes);
  at::cuda::CUDAGuard guard(pd.device());
  CHECK_LAUNCH(launch_multi_tensor_ema(
      cur_stream(), pd.data_ptr<int64_t>(), ps.data_ptr<int64_t>(),
      sizes.data_ptr<int32_t>(), pd.size(0), (float)m));
}

void multi_cast_bf16(torch::Tensor src_ptrs, torch::Tensor dst_ptrs,
                     torch::Tensor sizes) {
  CHECK_CUDA_CONTIG(src_ptrs);
  CHECK_CUDA_CONTIG(dst_ptrs);
  CHECK_CUDA_CONTIG(sizes);
  at::cuda::CUDAGuard guard(src_ptrs.device());
  CHECK_LAUNCH(launch_multi_cast_bf16(
      cur_stream(), src_ptrs.data_ptr<int64_t>(),
      dst_ptrs.data_ptr<int64_t>(), sizes.data_ptr<int32_t>(),
      src_ptrs.size(0)));
}

void multi_transpose_bf16(torch::Tensor src_ptrs, torch::Tensor dst_ptrs,
                          torch::Tensor meta) {
  CHECK_CUDA_CONTIG(src_ptrs);
  CHECK_CUDA_CONTIG(dst_ptrs);
  CHECK_CUDA_CONTIG(meta);
  TORCH_CHECK(meta.numel() == src_ptrs.size(0) * 4);
  at::cuda::CUDAGuard guard(src_ptrs.device());
  CHECK_LAUNCH(launch_multi_transpose_bf16(
      cur_stream(), src_ptrs.data_ptr<int64_t>(),
      dst_ptrs.data_ptr<int64_t>(), meta.data_ptr<int32_t>(),
      src_ptrs.size(0)));
}

void multi_tensor_sgd(torch::Tensor pp, torch::Tensor pg, torch::Tensor pb,
                      torch::Tensor sizes, double lr, double momentum,
                      double wd) {
  at::cuda::CUDAGuard guard(pp.device());
  CHECK_LAUNCH(launch_multi_tensor_sgd(
      cur_stream(), pp.data_ptr<int64_t>(), pg.data_ptr<int64_t>(),
      pb.data_ptr<int64_t>(), sizes.data_ptr<int32_t>(), pp.size(0),
      (float)lr, (float)momentum, (float)wd));
}

void ema_flat(torch::Tensor dst, torch::Tensor src, double m) {
  CHECK_CUDA_CONTIG(dst);
  CHECK_CUDA_CONTIG(src);
  TORCH_CHECK(dst.scalar_type() == torch::kFloat32 &&
              src.scalar_type() == torch::kFloat32 &&
              dst.numel() == src.numel());
  at::cuda::CUDAGuard guard(dst.device());
  CHECK_LAUNCH(launch_ema_flat(cur_stream(), dst.data_ptr<float>(),
                               src.data_ptr<float>(), dst.numel(),
                               (float)m));
}

// ---------------------------------------------------------------------------
// Fused BatchNorm (+ReLU, +residual).  x is a channels_last [N,C,H,W] bf16
// tensor viewed here as [R, C] with R = N*H*W.

std::vector<torch::Tensor> bn_fwd_train(torch::Tensor x,
                                        c10::optional<torch::Tensor> res,
                                        torch::Tensor gamma,
                                        torch::Tensor beta,
                                        c10::optional<torch::Tensor> rmean,
                                        c10::optional<torch::Tensor> rvar,
                                        double eps, double momentum,
                                        bool relu) {
  TORCH_CHECK(x.is_cuda() && is_bf16(x));
  const int64_t C = x.size(1);
  const int64_t R = x.numel() / C;
  at::cuda::CUDAGuard guard(x.device());
  auto f32 = x.options().dtype(torch::kFloat32);
  auto part_s = torch::empty({1024, C}, f32);
  auto part_q = torch::empty({1024, C}, f32);
  auto mean = torch::empty({C}, f32);
  auto rstd = torch::empty({C}, f32);
  CHECK_LAUNCH(launch_bn_stats(
      cur_stream(), x.data_ptr(), part_s.data_ptr<float>(),
      part_q.data_ptr<float>(), mean.data_ptr<float>(),
      rstd.data_ptr<float>(),
      rmean.has_value() ? rmean->data_ptr<float>() : nullptr,
      rvar.has_value() ? rvar->data_ptr<float>() : nullptr, R, C, (float)eps,
      (float)momentum));
  auto y = torch::empty_like(x);
  CHECK_LAUNCH(launch_bn_apply(
      cur_stream(), x.data_ptr(),
      res.has_value() ? res->data_ptr() : nullptr, y.data_ptr(),
      mean.data_ptr<float>(), rstd.data_ptr<float>(),
      gamma.data_ptr<float>(), beta.data_ptr<float>(), R, C, relu));
  return {y, mean, rstd};
}

torch::Tensor bn_fwd_eval(torch::Tensor x, c10::optional<torch::Tensor> res,
                          torch::Tensor gamma, torch::Tensor beta,
                          torch::Tensor mean, torch::Tensor rstd, bool relu) {
  TORCH_CHECK(x.is_cuda() && is_bf16(x));
  const int64_t C = x.size(1);
  const int64_t R = x.numel() / C;
  at::cuda::CUDAGuard guard(x.device());
  auto y = torch::empty_like(x);
  CHECK_LAUNCH(launch_bn_apply(
      cur_stream(), x.data_ptr(),
      res.has_value() ? res->data_ptr() : nullptr, y.data_ptr(),
      mean.data_ptr<float>(), rstd.data_ptr<float>(),
      gamma.data_ptr<float>(), beta.data_ptr<float>(), R, C, relu));
  return y;
}

std::vector<torch::Tensor> bn_bwd(torch::Tensor g, torch::Tensor y,
                                  torch::Tensor x, torch::Tensor mean,
                                  torch::Tensor rstd, torch::Tensor gamma,
                                  bool relu, bool train, bool need_dres) {
  TORCH_CHECK(g.is_cuda() && is_bf16(g));
  const int64_t C = x.size(1);
  const int64_t R = x.numel() / C;
  at::cuda::CUDAGuard guard(x.device());
  auto f32 = x.options().dtype(torch::kFloat32);
  auto part_db = torch::empty({1024, C}, f32);
  auto part_dg = torch::empty({1024, C}, f32);
  auto dbeta = torch::empty({C}, f32);
  auto dgamma = torch::empty({C}, f32);
  CHECK_LAUNCH(launch_bn_bwd_reduce(
      cur_stream(), g.data_ptr(), y.data_ptr(), x.data_ptr(),
      mean.data_ptr<float>(), rstd.data_ptr<float>(),
      part_db.data_ptr<float>(), part_dg.data_ptr<float>(),
      dbeta.data_ptr<float>(), dgamma.data_ptr<float>(), R, C, relu));
  auto dx = torch::empty_like(g);
  torch::Tensor dres;
  if (need_dres) dres = torch::empty_like(g);
  CHECK_LAUNCH(launch_bn_bwd_dx(
      cur_stream(), g.data_ptr(), y.data_ptr(), x.data_ptr(),
      mean.data_ptr<float>(), rstd.data_ptr<float>(),
      gamma.data_ptr<float>(), dbeta.data_ptr<float>(),
      dgamma.data_ptr<float>(), dx.data_ptr(),
      need_dres ? dres.data_ptr() : nullptr, R, C, relu, train));
  if (need_dres) return {dx, dgamma, dbeta, dres};
  return {dx, dgamma, dbeta};
}

// ---------------------------------------------------------------------------
// bf16 NT-GEMM (1x1-conv forward/dgrad): C[M,N] = A[M,K] @ Bt[N,K]^T.
// c_add (optional): epilogue C += c_add — fuses the residual-path gradient
// accumulation into the dgrad GEMM store tail.
torch::Tensor gemm_nt_bf16(torch::Tensor A, torch::Tensor Bt,
                           c10::optional<torch::Tensor> c_add) {
  CHECK_CUDA_CONTIG(A);
  CHECK_CUDA_CONTIG(Bt);
  TORCH_CHECK(is_bf16(A) && is_bf16(Bt) && A.size(1) == Bt.size(1));
  const int64_t M = A.size(0), K = A.size(1), N = Bt.size(0);
  if (c_add.has_value())
    TORCH_CHECK(is_bf16(*c_add) && c_add->is_contiguous() &&
                c_add->numel() == M * N);
  at::cuda::CUDAGuard guard(A.device());
  auto C = torch::empty({M, N}, A.options());
  // LDS-staged kernel for conforming shapes, naive fragment-load fallback
  CHECK_LAUNCH(launch_gemm_nt_lds_bf16(
      cur_stream(), A.data_ptr(), Bt.data_ptr(), C.data_ptr(), M, N, K,
      c_add.has_value() ? c_add->data_ptr() : nullptr));
  return C;
}

// GEMM with fused BN-apply (A path) / BN-stats (output) — the conv+BN hot
// path.  want_aout materializes the transformed A operand (consumer conv
// backward wgrad input); c_add fuses an epilogue add.
std::vector<torch::Tensor> gemm_nt_fused(
    torch::Tensor A, torch::Tensor Bt, c10::optional<torch::Tensor> a_scale,
    c10::optional<torch::Tensor> a_shift, bool a_relu, bool want_stats,
    bool want_aout, c10::optional<torch::Tensor> c_add,
    c10::optional<torch::Tensor> ws_sum,
    c10::optional<torch::Tensor> ws_ssq) {
  CHECK_CUDA_CONTIG(A);
  CHECK_CUDA_CONTIG(Bt);
  TORCH_CHECK(is_bf16(A) && is_bf16(Bt) && A.size(1) == Bt.size(1));
  const int64_t M = A.size(0), K = A.size(1), N = Bt.size(0);
  TORCH_CHECK(a_scale.has_value() == a_shift.has_value());
  if (a_scale.has_value())
    TORCH_CHECK(a_scale->numel() == K && a_shift->numel() == K);
  TORCH_CHECK(!want_aout || a_scale.has_value(),
              "want_aout requires the BN-apply prologue");
  if (c_add.has_value())
    TORCH_CHECK(is_bf16(*c_add) && c_add->is_contiguous() &&
                c_add->numel() == M * N);
  at::cuda::CUDAGuard guard(A.device());
  auto C = torch::empty({M, N}, A.options());
  torch::Tensor ssum, ssq, aout;
  if (want_stats) {
    // slot-striped [256, N] partials (see conv1x1.hip STAT_SLOTS);
    // bn_finalize reduces the slots.  When the caller provides a
    // workspace pair it must be pre-zeroed — the finalize(zero_src=True)
    // restores that invariant, so no fill kernels run per step.
    if (ws_sum.has_value()) {
      TORCH_CHECK(ws_sum->numel() == 256 * N && ws_sum->is_contiguous() &&
                  ws_ssq->numel() == 256 * N && ws_ssq->is_contiguous());
      ssum = *ws_sum;
      ssq = *ws_ssq;
    } else {
      ssum = torch::zeros({256, N}, A.options().dtype(torch::kFloat32));
      ssq = torch::zeros({256, N}, A.options().dtype(torch::kFloat32));
    }
  }
  if (want_aout) aout = torch::empty({M, K}, A.options());
  CHECK_LAUNCH(launch_gemm_nt_fused_bf16(
      cur_stream(), A.data_ptr(), Bt.data_ptr(), C.data_ptr(), M, N, K,
      a_scale.has_value() ? a_scale->data_ptr<float>() : nullptr,
      a_shift.has_value() ? a_shift->data_ptr<float>() : nullptr, a_relu,
      want_stats ? ssum.data_ptr<float>() : nullptr,
      want_stats ? ssq.data_ptr<float>() : nullptr,
      want_aout ? aout.data_ptr() : nullptr,
      c_add.has_value() ? c_add->data_ptr() : nullptr));
  std::vector<torch::Tensor> out = {C};
  if (want_stats) {
    out.push_back(ssum);
    out.push_back(ssq);
  }
  if (want_aout) out.push_back(aout);
  return out;
}

// fp8 e4m3 GEMM (MX-scaled MFMA, per-tensor software scaling):
// C_bf16 = (A8 @ B8t^T) * out_scale.  want_stats adds the bn-stats
// epilogue ([256, N] slot slabs like the bf16 path).
std::vector<torch::Tensor> gemm_nt_fp8(torch::Tensor A, torch::Tensor Bt,
                                       double out_scale, bool want_stats,
                                       c10::optional<torch::Tensor> ws_sum,
                                       c10::optional<torch::Tensor> ws_ssq) {
  CHECK_CUDA_CONTIG(A);
  CHECK_CUDA_CONTIG(Bt);
  TORCH_CHECK(A.scalar_type() == torch::kFloat8_e4m3fn &&
              Bt.scalar_type() == torch::kFloat8_e4m3fn &&
              A.size(1) == Bt.size(1));
  const int64_t M = A.size(0), K = A.size(1), N = Bt.size(0);
  at::cuda::CUDAGuard guard(A.device());
  auto C = torch::empty({M, N}, A.options().dtype(torch::kBFloat16));
  torch::Tensor ssum, ssq;
  if (want_stats) {
    if (ws_sum.has_value()) {
      ssum = *ws_sum;
      ssq = *ws_ssq;
    } else {
      ssum = torch::zeros({256, N}, A.options().dtype(torch::kFloat32));
      ssq = torch::zeros({256, N}, A.options().dtype(torch::kFloat32));
    }
  }
  CHECK_LAUNCH(launch_gemm_nt_fp8(
      cur_stream(), A.data_ptr(), Bt.data_ptr(), C.data_ptr(), M, N, K,
      (float)out_scale, want_stats ? ssum.data_ptr<float>() : nullptr,
      want_stats ? ssq.data_ptr<float>() : nullptr));
  if (want_stats) return {C, ssum, ssq};
  return {C};
}

// ---------------------------------------------------------------------------
// Granular BN entry points for the conv+BN fused block path.

// stats only (1 read): mean/rstd + running-stats update
std::vector<torch::Tensor> bn_stats(torch::Tensor x,
                                    c10::optional<torch::Tensor> rmean,
                                    c10::optional<torch::Tensor> rvar,
                                    double eps, double momentum) {
  TORCH_CHECK(x.is_cuda() && is_bf16(x));
  const int64_t C = x.size(1);
  const int64_t R = x.numel() / C;
  at::cuda::CUDAGuard guard(x.device());
  auto f32 = x.options().dtype(torch::kFloat32);
  auto part_s = torch::empty({1024, C}, f32);
  auto part_q = torch::empty({1024, C}, f32);
  auto mean = torch::empty({C}, f32);
  auto rstd = torch::empty({C}, f32);
  CHECK_LAUNCH(launch_bn_stats(
      cur_stream(), x.data_ptr(), part_s.data_ptr<float>(),
      part_q.data_ptr<float>(), mean.data_ptr<float>(),
      rstd.data_ptr<float>(),
      rmean.has_value() ? rmean->data_ptr<float>() : nullptr,
      rvar.has_value() ? rvar->data_ptr<float>() : nullptr, R, C, (float)eps,
      (float)momentum));
  return {mean, rstd};
}

// finalize from the conv-GEMM epilogue's slot-striped channel partials
// ([P, C]; 1-D input means P = 1)
std::vector<torch::Tensor> bn_finalize(torch::Tensor ssum, torch::Tensor ssq,
                                       int64_t R,
                                       c10::optional<torch::Tensor> rmean,
                                       c10::optional<torch::Tensor> rvar,
                                       double eps, double momentum,
                                       bool zero_src) {
  CHECK_CUDA_CONTIG(ssum);
  CHECK_CUDA_CONTIG(ssq);
  const int64_t C = ssum.dim() == 2 ? ssum.size(1) : ssum.numel();
  const int64_t P = ssum.dim() == 2 ? ssum.size(0) : 1;
  at::cuda::CUDAGuard guard(ssum.device());
  auto mean = torch::empty({C}, ssum.options());
  auto rstd = torch::empty({C}, ssum.options());
  CHECK_LAUNCH(launch_bn_finalize(
      cur_stream(), ssum.data_ptr<float>(), ssq.data_ptr<float>(),
      mean.data_ptr<float>(), rstd.data_ptr<float>(),
      rmean.has_value() ? rmean->data_ptr<float>() : nullptr,
      rvar.has_value() ? rvar->data_ptr<float>() : nullptr, R, C, P,
      (float)eps, (float)momentum, zero_src));
  return {mean, rstd};
}

// dx-only backward: upstream grad already relu-masked (by the dgrad-GEMM
// epilogue), dbeta/dgamma already reduced there — one elementwise pass.
torch::Tensor bn_bwd_dx_only(torch::Tensor gp, torch::Tensor z,
                             torch::Tensor mean, torch::Tensor rstd,
                             torch::Tensor gamma, torch::Tensor dbeta,
                             torch::Tensor dgamma) {
  TORCH_CHECK(gp.is_cuda() && is_bf16(gp) && is_bf16(z));
  const int64_t C = z.size(1);
  const int64_t R = z.numel() / C;
  at::cuda::CUDAGuard guard(z.device());
  auto dx = torch::empty_like(gp);
  CHECK_LAUNCH(launch_bn_bwd_dx(
      cur_stream(), gp.data_ptr(), gp.data_ptr(), z.data_ptr(),
      mean.data_ptr<float>(), rstd.data_ptr<float>(),
      gamma.data_ptr<float>(), dbeta.data_ptr<float>(),
      dgamma.data_ptr<float>(), dx.data_ptr(), nullptr, R, C,
      /*relu=*/false, /*train=*/true));
  return dx;
}

// Round-2 prototype: 3x3 pad-1 conv (stride 1/2) as implicit MFMA GEMM,
// with optional fused BN-apply (input path) / BN-stats (output).
std::vector<torch::Tensor> conv3x3_fwd(
    torch::Tensor x, torch::Tensor w, int64_t stride,
    c10::optional<torch::Tensor> a_scale, c10::optional<torch::Tensor> a_shift,
    bool a_relu, bool want_stats, c10::optional<torch::Tensor> neutral) {
  TORCH_CHECK(x.is_cuda() && is_bf16(x) && x.dim() == 4);
  TORCH_CHECK(is_bf16(w) && w.dim() == 4 && w.size(2) == 3 && w.size(3) == 3);
  TORCH_CHECK(x.is_contiguous(torch::MemoryFormat::ChannelsLast) &&
              w.is_contiguous(torch::MemoryFormat::ChannelsLast),
              "conv3x3_fwd wants channels_last x and w");
  const int64_t N = x.size(0), Ci = x.size(1), H = x.size(2), W = x.size(3);
  const int64_t Co = w.size(0);
  TORCH_CHECK(w.size(1) == Ci);
  const int64_t Ho = (H - 1) / stride + 1, Wo = (W - 1) / stride + 1;
  at::cuda::CUDAGuard guard(x.device());
  auto out = torch::empty({N, Co, Ho, Wo}, x.options())
                 .contiguous(torch::MemoryFormat::ChannelsLast);
  torch::Tensor zp;
  if (neutral.has_value()) {
    TORCH_CHECK(is_bf16(*neutral) && neutral->numel() >= Ci + 8);
    zp = *neutral;
  } else {
    zp = torch::zeros({Ci + 8}, x.options());
  }
  torch::Tensor ssum, ssq;
  if (want_stats) {
    ssum = torch::zeros({Co}, x.options().dtype(torch::kFloat32));
    ssq = torch::zeros({Co}, x.options().dtype(torch::kFloat32));
  }
  CHECK_LAUNCH(launch_conv3x3_fwd(
      cur_stream(), x.data_ptr(), w.data_ptr(), zp.data_ptr(),
      out.data_ptr(), N, H, W, Ci, Co, stride,
      a_scale.has_value() ? a_scale->data_ptr<float>() : nullptr,
      a_shift.has_value() ? a_shift->data_ptr<float>() : nullptr, a_relu,
      want_stats ? ssum.data_ptr<float>() : nullptr,
      want_stats ? ssq.data_ptr<float>() : nullptr));
  if (want_stats) return {out, ssum, ssq};
  return {out};
}

torch::Tensor conv1x1_strided(torch::Tensor x, torch::Tensor w,
                              int64_t stride) {
  TORCH_CHECK(x.is_cuda() && is_bf16(x) && x.dim() == 4);
  TORCH_CHECK(is_bf16(w) && w.dim() == 4 && w.size(2) == 1 && w.size(3) == 1);
  TORCH_CHECK(x.is_contiguous(torch::MemoryFormat::ChannelsLast));
  const int64_t N = x.size(0), Ci = x.size(1), H = x.size(2), W = x.size(3);
  const int64_t Co = w.size(0);
  TORCH_CHECK(w.size(1) == Ci);
  const int64_t Ho = (H - 1) / stride + 1, Wo = (W - 1) / stride + 1;
  at::cuda::CUDAGuard guard(x.device());
  auto out = torch::empty({N, Co, Ho, Wo}, x.options())
                 .contiguous(torch::MemoryFormat::ChannelsLast);
  CHECK_LAUNCH(launch_conv1x1_strided(cur_stream(), x.data_ptr(),
                                      w.contiguous().data_ptr(),
                                      out.data_ptr(), N, H, W, Ci, Co,
                                      stride));
  return out;
}

// Round-2 prototype: dgrad GEMM + fused BN-backward epilogue.
std::vector<torch::Tensor> gemm_nt_dgrad_epi(torch::Tensor A,
                                             torch::Tensor Bt,
                                             torch::Tensor y, torch::Tensor z,
                                             torch::Tensor mean,
                                             torch::Tensor rstd,
                                             c10::optional<torch::Tensor>
                                                 c_add,
                                             c10::optional<torch::Tensor>
                                                 ws_db,
                                             c10::optional<torch::Tensor>
                                                 ws_dg) {
  CHECK_CUDA_CONTIG(A);
  CHECK_CUDA_CONTIG(Bt);
  TORCH_CHECK(is_bf16(A) && is_bf16(Bt) && is_bf16(y) && is_bf16(z));
  const int64_t M = A.size(0), K = A.size(1), N = Bt.size(0);
  TORCH_CHECK(y.numel() == M * N && z.numel() == M * N);
  TORCH_CHECK(mean.numel() == N && rstd.numel() == N);
  if (c_add.has_value())
    TORCH_CHECK(is_bf16(*c_add) && c_add->is_contiguous() &&
                c_add->numel() == M * N);
  at::cuda::CUDAGuard guard(A.device());
  auto C = torch::empty({M, N}, A.options());
  // slot-striped partials (conv1x1.hip STAT_SLOTS), reduced by the
  // self-zeroing slot_reduce2 (caller workspace must be pre-zeroed)
  torch::Tensor dbeta, dgamma;
  if (ws_db.has_value()) {
    TORCH_CHECK(ws_db->numel() == 256 * N && ws_db->is_contiguous() &&
                ws_dg->numel() == 256 * N && ws_dg->is_contiguous());
    dbeta = *ws_db;
    dgamma = *ws_dg;
  } else {
    dbeta = torch::zeros({256, N}, A.options().dtype(torch::kFloat32));
    dgamma = torch::zeros({256, N}, A.options().dtype(torch::kFloat32));
  }
  CHECK_LAUNCH(launch_gemm_nt_dgrad_epi(
      cur_stream(), A.data_ptr(), Bt.data_ptr(), C.data_ptr(), M, N, K,
      y.data_ptr(), z.data_ptr(), mean.data_ptr<float>(),
      rstd.data_ptr<float>(), dbeta.data_ptr<float>(),
      dgamma.data_ptr<float>(),
      c_add.has_value() ? c_add->data_ptr() : nullptr));
  auto db = torch::empty({N}, A.options().dtype(torch::kFloat32));
  auto dg = torch::empty({N}, A.options().dtype(torch::kFloat32));
  CHECK_LAUNCH(launch_slot_reduce2(
      cur_stream(), dbeta.data_ptr<float>(), dgamma.data_ptr<float>(),
      db.data_ptr<float>(), dg.data_ptr<float>(), N, 256));
  return {C, db, dg};
}

// ---------------------------------------------------------------------------
// MaxPool2d 3x3/s2/p1 (ResNet stem), NHWC bf16
std::vector<torch::Tensor> maxpool_fwd(torch::Tensor x) {
  TORCH_CHECK(x.is_cuda() && is_bf16(x) && x.dim() == 4);
  const int64_t N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  const int64_t Ho = (H + 1) / 2, Wo = (W + 1) / 2;
  at::cuda::CUDAGuard guard(x.device());
  auto y = torch::empty({N, C, Ho, Wo}, x.options())
               .contiguous(torch::MemoryFormat::ChannelsLast);
  auto idx = torch::empty({N, Ho, Wo, C}, x.options().dtype(torch::kUInt8));
  CHECK_LAUNCH(launch_maxpool_fwd(cur_stream(), x.data_ptr(), y.data_ptr(),
                                  idx.data_ptr(), N, C, H, W, Ho, Wo));
  return {y, idx};
}

// head adaptive-avg-pool to 1x1 (NHWC bf16)
torch::Tensor avgpool_fwd(torch::Tensor x) {
  TORCH_CHECK(x.is_cuda() && is_bf16(x) && x.dim() == 4);
  const int64_t N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  at::cuda::CUDAGuard guard(x.device());
  auto y = torch::empty({N, C, 1, 1}, x.options())
               .contiguous(torch::MemoryFormat::ChannelsLast);
  CHECK_LAUNCH(launch_avgpool_fwd(cur_stream(), x.data_ptr(), y.data_ptr(),
                                  N, C, H * W));
  return y;
}

torch::Tensor avgpool_bwd(torch::Tensor g, int64_t H, int64_t W) {
  TORCH_CHECK(g.is_cuda() && is_bf16(g));
  const int64_t N = g.size(0), C = g.size(1);
  at::cuda::CUDAGuard guard(g.device());
  auto dx = torch::empty({N, C, H, W}, g.options())
                .contiguous(torch::MemoryFormat::ChannelsLast);
  CHECK_LAUNCH(launch_avgpool_bwd(cur_stream(), g.data_ptr(), dx.data_ptr(),
                                  N, C, H * W));
  return dx;
}

torch::Tensor maxpool_bwd(torch::Tensor g, torch::Tensor idx, int64_t H,
                          int64_t W) {
  TORCH_CHECK(g.is_cuda() && is_bf16(g) && g.dim() == 4);
  const int64_t N = g.size(0), C = g.size(1), Ho = g.size(2), Wo = g.size(3);
  at::cuda::CUDAGuard guard(g.device());
  auto dx = torch::empty({N, C, H, W}, g.options())
                .contiguous(torch::MemoryFormat::ChannelsLast);
  CHECK_LAUNCH(launch_maxpool_bwd(cur_stream(), g.data_ptr(), idx.data_ptr(),
                                  dx.data_ptr(), N, C, H, W, Ho, Wo));
  return dx;
}

// ---------------------------------------------------------------------------
torch::Tensor two_crop_augment(torch::Tensor src, torch::Tensor params,
                               int64_t S, bool out_bf16, bool channels_last) {
  CHECK_CUDA_CONTIG(src);
  CHECK_CUDA_CONTIG(params);
  TORCH_CHECK((src.scalar_type() == torch::kFloat32 ||
               src.scalar_type() == torch::kUInt8) &&
              src.dim() == 4 && src.size(1) == 3,
              "src must be fp32 or uint8 [B,3,H,W]");
  TORCH_CHECK(params.scalar_type() == torch::kFloat32 &&
              params.size(1) == 12);
  const int64_t B = src.size(0), H = src.size(2), W = src.size(3);
  const int64_t M = params.size(0);
  at::cuda::CUDAGuard guard(src.device());
  // workspaces/out are fp32/bf16 regardless of the (possibly uint8) source
  auto f32 = src.options().dtype(torch::kFloat32);
  auto out_opts = out_bf16 ? f32.dtype(torch::kBFloat16) : f32;
  torch::Tensor out = torch::empty({M, 3, S, S}, out_opts);
  if (channels_last)
    out = out.contiguous(torch::MemoryFormat::ChannelsLast);
  auto tmp1 = torch::empty({M, 3, S, S}, f32);
  auto tmp2 = torch::empty({M, 3, S, S}, f32);
  auto graysum = torch::zeros({M}, f32);
  CHECK_LAUNCH(launch_two_crop_augment(
      cur_stream(), src.data_ptr(),
      src.scalar_type() == torch::kUInt8, params.data_ptr<float>(),
      out.data_ptr(), tmp1.data_ptr<float>(), tmp2.data_ptr<float>(),
      graysum.data_ptr<float>(), B, M, H, W, S, out_bf16, channels_last));
  return out;
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, mod) {
  mod.def("l2norm_rows_fwd", &l2norm_rows_fwd);
  mod.def("l2norm_rows_bwd", &l2norm_rows_bwd);
  mod.def("contrastive_fwd", &contrastive_fwd);
  mod.def("contrastive_bwd", &contrastive_bwd);
  mod.def("softmax_ce0_fwd", &softmax_ce0_fwd);
  mod.def("softmax_ce0_bwd", &softmax_ce0_bwd, py::arg("logits"),
          py::arg("row_max"), py::arg("row_lse"), py::arg("scale"),
          py::arg("gscale") = c10::nullopt);
  mod.def("queue_enqueue", &queue_enqueue);
  mod.def("queue_enqueue_dev", &queue_enqueue_dev);
  mod.def("multi_tensor_ema", &multi_tensor_ema);
  mod.def("multi_cast_bf16", &multi_cast_bf16);
  mod.def("multi_transpose_bf16", &multi_transpose_bf16);
  mod.def("multi_tensor_sgd", &multi_tensor_sgd);
  mod.def("ema_flat", &ema_flat);
  mod.def("two_crop_augment", &two_crop_augment);
  mod.def("bn_fwd_train", &bn_fwd_train);
  mod.def("bn_fwd_eval", &bn_fwd_eval);
  mod.def("bn_bwd", &bn_bwd);
  mod.def("bn_stats", &bn_stats);
  mod.def("bn_finalize", &bn_finalize, py::arg("ssum"), py::arg("ssq"),
          py::arg("R"), py::arg("rmean"), py::arg("rvar"), py::arg("eps"),
          py::arg("momentum"), py::arg("zero_src") = false);
  mod.def("bn_bwd_dx_only", &bn_bwd_dx_only);
  mod.def("maxpool_fwd", &maxpool_fwd);
  mod.def("gemm_nt_bf16", &gemm_nt_bf16, py::arg("A"), py::arg("Bt"),
          py::arg("c_add") = c10::nullopt);
  mod.def("gemm_nt_fp8", &gemm_nt_fp8, py::arg("A"), py::arg("Bt"),
          py::arg("out_scale"), py::arg("want_stats") = false,
          py::arg("ws_sum") = c10::nullopt, py::arg("ws_ssq") = c10::nullopt);
  mod.def("gemm_nt_fused", &gemm_nt_fused, py::arg("A"), py::arg("Bt"),
          py::arg("a_scale"), py::arg("a_shift"), py::arg("a_relu"),
          py::arg("want_stats"), py::arg("want_aout") = false,
          py::arg("c_add") = c10::nullopt,
          py::arg("ws_sum") = c10::nullopt,
          py::arg("ws_ssq") = c10::nullopt);
  mod.def("conv3x3_fwd", &conv3x3_fwd);
  mod.def("conv1x1_strided", &conv1x1_strided);
  mod.def("gemm_nt_dgrad_epi", &gemm_nt_dgrad_epi, py::arg("A"),
          py::arg("Bt"), py::arg("y"), py::arg("z"), py::arg("mean"),
          py::arg("rstd"), py::arg("c_add") = c10::nullopt,
          py::arg("ws_db") = c10::nullopt, py::arg("ws_dg") = c10::nullopt);
  mod.def("maxpool_bwd", &maxpool_bwd);
  mod.def("avgpool_fwd", &avgpool_fwd);
  mod.def("avgpool_bwd", &avgpool_bwd);
}
