// Python bindings for the machin_amd gfx950 kernels — written
// directly against PyTorch-ROCm's native HIP API (the
// "masquerading-as-CUDA" device layer every ROCm extension runs on),
// so the tree carries NO hipify-generated sources.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <ATen/hip/impl/HIPGuardImplMasqueradingAsCUDA.h>

#include <cstdint>
#include <vector>

using at::Tensor;

// launchers defined in the .hip translation units
void sumtree_update_launch(float*, const int64_t*, const float*, int64_t,
                           int64_t, int, hipStream_t);
void sumtree_build_launch(float*, int64_t, hipStream_t);
void sumtree_sample_launch(const float*, const float*, int64_t*, int64_t,
                           int64_t, int, int64_t, hipStream_t);
void discounted_returns_launch(const float*, const float*, const float*,
                               float*, int64_t, int64_t, float, hipStream_t);
void gae_launch(const float*, const float*, const float*, const float*,
                float*, int64_t, int64_t, float, float, hipStream_t);
void vtrace_launch(const float*, const float*, const float*, const float*,
                   const float*, const float*, float*, float*, int64_t,
                   int64_t, float, float, float, float, hipStream_t);
void nstep_returns_launch(const float*, const float*, float*, int64_t,
                          int64_t, float, int, hipStream_t);
void vtrace_bt_launch(const float*, const float*, const float*,
                      const float*, const float*, const float*, float*,
                      float*, int64_t, int64_t, float, float, float,
                      float, hipStream_t);
void categorical_projection_launch(const float*, const float*, const float*,
                                   float*, int64_t, int64_t, float, float,
                                   float, hipStream_t);
void multi_tensor_polyak_launch(void*, const int64_t*, int64_t, int64_t,
                                float, hipStream_t);
void fused_rmsprop_launch(void*, const int64_t*, int64_t, int64_t,
                          float*, float, float, float, float,
                          hipStream_t);
void gaussian_sample_logprob_launch(const float*, const float*, float*,
                                    float*, int64_t, int64_t, uint64_t,
                                    uint64_t, int, float, hipStream_t);
void gaussian_logprob_launch(const float*, const float*, const float*,
                             float*, int64_t, int64_t, int, float,
                             hipStream_t);
void normal_noise_launch(float*, int64_t, float, float, uint64_t, uint64_t,
                         int, hipStream_t);
void ou_update_launch(float*, int64_t, float, float, float, float, uint64_t,
                      uint64_t, hipStream_t);
void u8_to_bf16_scale_launch(const unsigned char*, void*, int64_t, float,
                             hipStream_t);
void pg_head_fwd_launch(const void*, const int64_t*, float*, float*,
                        int64_t, int, hipStream_t);
void pg_head_bwd_launch(const void*, const int64_t*, const float*,
                        const float*, void*, int64_t, int, hipStream_t);
void conv1_wrw_launch(const void*, const unsigned char*, float*, float*,
                      float*, int64_t, float, hipStream_t);
void mfma_probe_launch(const void*, const void*, float*, hipStream_t);
void tr16_probe_launch(float*, int, hipStream_t);
void fwd_bfrag_probe_launch(float*, int, hipStream_t);
void tr16_diag_launch(float*, int, hipStream_t);
void conv1_fwd_launch(const unsigned char*, const void*, const float*,
                      void*, int64_t, float, hipStream_t);

namespace {

void check_f32_cuda(const Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be a CUDA tensor");
  TORCH_CHECK(t.scalar_type() == at::kFloat, name, " must be float32");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

hipStream_t current_stream() {
  return at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
}

// ------------------------------------------------------------------
// sum-tree
// ------------------------------------------------------------------
void sumtree_update(Tensor tree, Tensor idx, Tensor w, int64_t capacity,
                    int64_t depth) {
  check_f32_cuda(tree, "tree");
  check_f32_cuda(w, "w");
  TORCH_CHECK(idx.is_cuda() && idx.scalar_type() == at::kLong &&
                  idx.is_contiguous(),
              "idx must be contiguous int64 CUDA");
  const at::hip::OptionalHIPGuardMasqueradingAsCUDA guard(tree.device());
  sumtree_update_launch(tree.data_ptr<float>(), idx.data_ptr<int64_t>(),
                        w.data_ptr<float>(), idx.numel(), capacity,
                        (int)depth, current_stream());
}

void sumtree_build(Tensor tree, int64_t capacity) {
  check_f32_cuda(tree, "tree");
  const at::hip::OptionalHIPGuardMasqueradingAsCUDA guard(tree.device());
  sumtree_build_launch(tree.data_ptr<float>(), capacity, current_stream());
}

Tensor sumtree_sample(Tensor tree, Tensor u, int64_t capacity, int64_t depth,
                      int64_t size) {
  check_f32_cuda(tree, "tree");
  check_f32_cuda(u, "u");
  const at::hip::OptionalHIPGuardMasqueradingAsCUDA guard(tree.device());
  Tensor out = at::empty({u.numel()}, u.options().dtype(at::kLong));
  sumtree_sample_launch(tree.data_ptr<float>(), u.data_ptr<float>(),
                        out.data_ptr<int64_t>(), u.numel(), capacity,
                        (int)depth, size, current_stream());
  return out;
}

// ------------------------------------------------------------------
// scans
// ------------------------------------------------------------------
Tensor discounted_returns(Tensor rew, Tensor nd, Tensor bootstrap,
                          double gamma) {
  check_f32_cuda(rew, "rewards");
  const at::hip::OptionalHIPGuardMasqueradingAsCUDA guard(rew.device());
  int64_t T = rew.size(0), B = rew.size(1);
  Tensor out = at::empty_like(rew);
  discounted_returns_launch(rew.data_ptr<float>(), nd.data_ptr<float>(),
                            bootstrap.data_ptr<float>(),
                            out.data_ptr<float>(), T, B, (float)gamma,
                            current_stream());
  return out;
}

Tensor gae(Tensor rew, Tensor val, Tensor next_val, Tensor nd, double gamma,
           double lam) {
  check_f32_cuda(rew, "rewards");
  const at::hip::OptionalHIPGuardMasqueradingAsCUDA guard(rew.device());
  int64_t T = rew.size(0), B = rew.size(1);
  Tensor out = at::empty_like(rew);
  gae_launch(rew.data_ptr<float>(), val.data_ptr<float>(),
             next_val.data_ptr<float>(), nd.data_ptr<float>(),
             out.data_ptr<float>(), T, B, (float)gamma, (float)lam,
             current_stream());
  return out;
}

std::vector<Tensor> vtrace(Tensor blp, Tensor tlp, Tensor rew, Tensor val,
                           Tensor bootstrap, Tensor nd, double gamma,
                           double rho_clip, double c_clip,
                           double pg_rho_clip) {
  check_f32_cuda(rew, "rewards");
  const at::hip::OptionalHIPGuardMasqueradingAsCUDA guard(rew.device());
  int64_t T = rew.size(0), B = rew.size(1);
  Tensor vs = at::empty_like(rew);
  Tensor pg_adv = at::empty_like(rew);
  vtrace_launch(blp.data_ptr<float>(), tlp.data_ptr<float>(),
                rew.data_ptr<float>(), val.data_ptr<float>(),
                bootstrap.data_ptr<float>(), nd.data_ptr<float>(),
                vs.data_ptr<float>(), pg_adv.data_ptr<float>(), T, B,
                (float)gamma, (float)rho_clip, (float)c_clip,
                (float)pg_rho_clip, current_stream());
  return {vs, pg_adv};
}

std::vector<Tensor> vtrace_bt(Tensor blp, Tensor tlp, Tensor rew,
                              Tensor val, Tensor bootstrap, Tensor nd,
                              double gamma, double rho_clip,
                              double c_clip, double pg_rho_clip) {
  check_f32_cuda(rew, "rewards");
  const at::hip::OptionalHIPGuardMasqueradingAsCUDA guard(rew.device());
  int64_t B = rew.size(0), T = rew.size(1);
  Tensor vs = at::empty_like(rew);
  Tensor pg_adv = at::empty_like(rew);
  vtrace_bt_launch(blp.data_ptr<float>(), tlp.data_ptr<float>(),
                   rew.data_ptr<float>(), val.data_ptr<float>(),
                   bootstrap.data_ptr<float>(), nd.data_ptr<float>(),
                   vs.data_ptr<float>(), pg_adv.data_ptr<float>(), T, B,
                   (float)gamma, (float)rho_clip, (float)c_clip,
                   (float)pg_rho_clip, current_stream());
  return {vs, pg_adv};
}

Tensor nstep_returns(Tensor rew, Tensor alive, double gamma, int64_t n) {
  check_f32_cuda(rew, "rewards");
  check_f32_cuda(alive, "alive");
  const at::hip::OptionalHIPGuardMasqueradingAsCUDA guard(rew.device());
  int64_t T = rew.size(0), B = rew.size(1);
  Tensor out = at::empty_like(rew);
  nstep_returns_launch(rew.data_ptr<float>(), alive.data_ptr<float>(),
                       out.data_ptr<float>(), T, B, (float)gamma, (int)n,
                       current_stream());
  return out;
}

// ------------------------------------------------------------------
// categorical projection
// ------------------------------------------------------------------
Tensor categorical_projection(Tensor next_dist, Tensor rew, Tensor nd,
                              double gamma, double v_min, double v_max) {
  check_f32_cuda(next_dist, "next_dist");
  const at::hip::OptionalHIPGuardMasqueradingAsCUDA guard(next_dist.device());
  int64_t B = next_dist.size(0), A = next_dist.size(1);
  TORCH_CHECK(A <= 256, "categorical projection supports at most 256 atoms");
  Tensor out = at::empty_like(next_dist);
  categorical_projection_launch(next_dist.data_ptr<float>(),
                                rew.data_ptr<float>(), nd.data_ptr<float>(),
                                out.data_ptr<float>(), B, A, (float)gamma,
                                (float)v_min, (float)v_max,
                                current_stream());
  return out;
}

// ------------------------------------------------------------------
// multi-tensor polyak
// ------------------------------------------------------------------
void multi_tensor_polyak(std::vector<Tensor> targets,
                         std::vector<Tensor> sources, double tau) {
  TORCH_CHECK(targets.size() == sources.size(), "list size mismatch");
  if (targets.empty()) return;
  const at::hip::OptionalHIPGuardMasqueradingAsCUDA guard(targets[0].device());
  int64_t n = (int64_t)targets.size();
  // host-side table: [tgt_ptr, src_ptr] pairs then exclusive prefix
  Tensor table = at::empty({n * 2 + n + 1},
                           at::TensorOptions().dtype(at::kLong));
  int64_t* h = table.data_ptr<int64_t>();
  int64_t total = 0;
  for (int64_t i = 0; i < n; ++i) {
    check_f32_cuda(targets[i], "target");
    check_f32_cuda(sources[i], "source");
    TORCH_CHECK(targets[i].numel() == sources[i].numel(), "numel mismatch");
    h[2 * i] = (int64_t)targets[i].data_ptr<float>();
    h[2 * i + 1] = (int64_t)sources[i].data_ptr<float>();
    h[2 * n + i] = total;
    total += targets[i].numel();
  }
  h[2 * n + n] = total;
  Tensor dev_table = table.to(targets[0].device(), /*non_blocking=*/true);
  int64_t* d = dev_table.data_ptr<int64_t>();
  multi_tensor_polyak_launch((void*)d, d + 2 * n, n, total, (float)tau,
                             current_stream());
}

// Cached variant: the python layer builds the device table ONCE per
// (targets, sources) pair; each call is then a single kernel launch
// with zero host-side setup (the v1 per-call table build made the
// fused kernel 6x slower than torch _foreach on small nets,
// profiles/kernel_bench_r01.json).
void multi_tensor_polyak_cached(Tensor dev_table, int64_t n_tensors,
                                int64_t total, double tau) {
  TORCH_CHECK(dev_table.is_cuda() &&
                  dev_table.scalar_type() == at::kLong &&
                  dev_table.is_contiguous(),
              "dev_table must be contiguous int64 CUDA");
  const at::hip::OptionalHIPGuardMasqueradingAsCUDA guard(
      dev_table.device());
  int64_t* d = dev_table.data_ptr<int64_t>();
  multi_tensor_polyak_launch((void*)d, d + 2 * n_tensors, n_tensors, total,
                             (float)tau, current_stream());
}

// Cached-plan fused clip+RMSprop: dev_table = [p,g,sq]*n ptrs then
// the n+1 prefix; norm_buf = one fp32 scratch.
void fused_rmsprop_cached(Tensor dev_table, int64_t n_tensors,
                          int64_t total, Tensor norm_buf,
                          double max_norm, double lr, double alpha,
                          double eps) {
  TORCH_CHECK(dev_table.is_cuda() &&
                  dev_table.scalar_type() == at::kLong &&
                  dev_table.is_contiguous(),
              "dev_table must be contiguous int64 CUDA");
  check_f32_cuda(norm_buf, "norm_buf");
  const at::hip::OptionalHIPGuardMasqueradingAsCUDA guard(
      dev_table.device());
  int64_t* d = dev_table.data_ptr<int64_t>();
  fused_rmsprop_launch((void*)d, d + 3 * n_tensors, n_tensors, total,
                       norm_buf.data_ptr<float>(), (float)max_norm,
                       (float)lr, (float)alpha, (float)eps,
                       current_stream());
}

// ------------------------------------------------------------------
// distributions
// ------------------------------------------------------------------
std::vector<Tensor> gaussian_sample_logprob(Tensor mu, Tensor log_std,
                                            int64_t seed, int64_t offset,
                                            bool tanh_squash,
                                            double epsilon) {
  check_f32_cuda(mu, "mu");
  check_f32_cuda(log_std, "log_std");
  const at::hip::OptionalHIPGuardMasqueradingAsCUDA guard(mu.device());
  int64_t B = mu.size(0), D = mu.size(1);
  Tensor act = at::empty_like(mu);
  Tensor logp = at::empty({B, 1}, mu.options());
  gaussian_sample_logprob_launch(
      mu.data_ptr<float>(), log_std.data_ptr<float>(), act.data_ptr<float>(),
      logp.data_ptr<float>(), B, D, (uint64_t)seed, (uint64_t)offset,
      tanh_squash ? 1 : 0, (float)epsilon, current_stream());
  return {act, logp};
}

Tensor gaussian_logprob(Tensor mu, Tensor log_std, Tensor act,
                        bool tanh_squash, double epsilon) {
  check_f32_cuda(mu, "mu");
  const at::hip::OptionalHIPGuardMasqueradingAsCUDA guard(mu.device());
  int64_t B = mu.size(0), D = mu.size(1);
  Tensor logp = at::empty({B, 1}, mu.options());
  gaussian_logprob_launch(mu.data_ptr<float>(), log_std.data_ptr<float>(),
                          act.data_ptr<float>(), logp.data_ptr<float>(), B,
                          D, tanh_squash ? 1 : 0, (float)epsilon,
                          current_stream());
  return logp;
}

void normal_noise_(Tensor x, double mean, double std, int64_t seed,
                   int64_t offset, bool add) {
  check_f32_cuda(x, "x");
  const at::hip::OptionalHIPGuardMasqueradingAsCUDA guard(x.device());
  normal_noise_launch(x.data_ptr<float>(), x.numel(), (float)mean,
                      (float)std, (uint64_t)seed, (uint64_t)offset,
                      add ? 1 : 0, current_stream());
}

void ou_update_(Tensor x, double mu, double theta, double sigma, double dt,
                int64_t seed, int64_t offset) {
  check_f32_cuda(x, "x");
  const at::hip::OptionalHIPGuardMasqueradingAsCUDA guard(x.device());
  ou_update_launch(x.data_ptr<float>(), x.numel(), (float)mu, (float)theta,
                   (float)sigma, (float)dt, (uint64_t)seed,
                   (uint64_t)offset, current_stream());
}

std::vector<Tensor> pg_head_fwd(Tensor logits, Tensor actions) {
  TORCH_CHECK(logits.is_cuda() && logits.scalar_type() == at::kBFloat16 &&
                  logits.is_contiguous(),
              "logits must be contiguous bf16 CUDA [N, A]");
  TORCH_CHECK(actions.scalar_type() == at::kLong && actions.is_contiguous(),
              "actions must be contiguous int64");
  int64_t N = logits.size(0), A = logits.size(1);
  TORCH_CHECK(A <= 32, "pg_head supports at most 32 actions");
  const at::hip::OptionalHIPGuardMasqueradingAsCUDA guard(logits.device());
  Tensor tl = at::empty({N}, logits.options().dtype(at::kFloat));
  Tensor ent = at::empty({N}, logits.options().dtype(at::kFloat));
  pg_head_fwd_launch(logits.data_ptr(), actions.data_ptr<int64_t>(),
                     tl.data_ptr<float>(), ent.data_ptr<float>(), N,
                     (int)A, current_stream());
  return {tl, ent};
}

Tensor pg_head_bwd(Tensor logits, Tensor actions, Tensor g_taken,
                   Tensor g_ent) {
  int64_t N = logits.size(0), A = logits.size(1);
  check_f32_cuda(g_taken, "g_taken");
  check_f32_cuda(g_ent, "g_ent");
  const at::hip::OptionalHIPGuardMasqueradingAsCUDA guard(logits.device());
  Tensor d = at::empty_like(logits);
  pg_head_bwd_launch(logits.data_ptr(), actions.data_ptr<int64_t>(),
                     g_taken.data_ptr<float>(), g_ent.data_ptr<float>(),
                     d.data_ptr(), N, (int)A, current_stream());
  return d;
}

Tensor u8_to_bf16_scale(Tensor in, double scale) {
  TORCH_CHECK(in.is_cuda() && in.scalar_type() == at::kByte &&
                  in.is_contiguous(),
              "input must be contiguous uint8 CUDA");
  const at::hip::OptionalHIPGuardMasqueradingAsCUDA guard(in.device());
  Tensor out = at::empty_like(in, in.options().dtype(at::kBFloat16));
  u8_to_bf16_scale_launch(in.data_ptr<unsigned char>(), out.data_ptr(),
                          in.numel(), (float)scale, current_stream());
  return out;
}

std::vector<Tensor> conv1_wrw(Tensor dy, Tensor frames, double scale) {
  // dy: [K, 32] bf16 (NHWC-flattened conv output grad);
  // frames: [B, 84, 84, 4] u8. Returns [32, 4, 8, 8] fp32 grad.
  TORCH_CHECK(dy.is_cuda() && dy.scalar_type() == at::kBFloat16 &&
                  dy.is_contiguous() && dy.size(1) == 32,
              "dy must be contiguous [K,32] bf16 CUDA");
  TORCH_CHECK(frames.is_cuda() && frames.scalar_type() == at::kByte &&
                  frames.is_contiguous(),
              "frames must be contiguous u8 CUDA");
  TORCH_CHECK(dy.size(0) == frames.size(0) * 400,
              "K must equal batch*400");
  const at::hip::OptionalHIPGuardMasqueradingAsCUDA guard(dy.device());
  Tensor scratch = at::empty({32 * 256},
                             dy.options().dtype(at::kFloat));
  Tensor grad_w = at::empty({32, 4, 8, 8},
                            dy.options().dtype(at::kFloat));
  Tensor grad_b = at::empty({32}, dy.options().dtype(at::kFloat));
  conv1_wrw_launch(dy.data_ptr(), frames.data_ptr<unsigned char>(),
                   scratch.data_ptr<float>(), grad_w.data_ptr<float>(),
                   grad_b.data_ptr<float>(), dy.size(0), (float)scale,
                   current_stream());
  return {grad_w, grad_b};
}

Tensor conv1_fwd(Tensor frames, Tensor weight, Tensor bias,
                 double scale) {
  // frames: [B, 84, 84, 4] u8 NHWC; weight: [256, 32] bf16 (patch-major
  // repack of the conv weight); bias: [32] fp32 or empty.
  TORCH_CHECK(frames.is_cuda() && frames.scalar_type() == at::kByte &&
                  frames.is_contiguous(),
              "frames must be contiguous u8 CUDA");
  TORCH_CHECK(weight.scalar_type() == at::kBFloat16 &&
                  weight.is_contiguous() && weight.size(0) == 256 &&
                  weight.size(1) == 32,
              "weight must be contiguous [256,32] bf16");
  const at::hip::OptionalHIPGuardMasqueradingAsCUDA guard(frames.device());
  int64_t K = frames.size(0) * 400;
  Tensor out = at::empty({K, 32},
                         weight.options().dtype(at::kBFloat16));
  const float* bias_ptr = nullptr;
  if (bias.defined() && bias.numel() == 32) {
    TORCH_CHECK(bias.scalar_type() == at::kFloat && bias.is_contiguous(),
                "bias must be contiguous fp32");
    bias_ptr = bias.data_ptr<float>();
  }
  conv1_fwd_launch(frames.data_ptr<unsigned char>(), weight.data_ptr(),
                   bias_ptr, out.data_ptr(), K, (float)scale,
                   current_stream());
  return out;
}

Tensor tr16_probe(int64_t mode) {
  Tensor out = at::empty({64, 4}, at::TensorOptions()
                                      .dtype(at::kFloat)
                                      .device(at::kCUDA));
  const at::hip::OptionalHIPGuardMasqueradingAsCUDA guard(out.device());
  tr16_probe_launch(out.data_ptr<float>(), (int)mode, current_stream());
  return out;
}

Tensor fwd_bfrag_probe(int64_t which) {
  Tensor out = at::empty({2, 256, 8}, at::TensorOptions()
                                          .dtype(at::kFloat)
                                          .device(at::kCUDA));
  const at::hip::OptionalHIPGuardMasqueradingAsCUDA guard(out.device());
  fwd_bfrag_probe_launch(out.data_ptr<float>(), (int)which,
                         current_stream());
  return out;
}

Tensor tr16_diag(int64_t mode) {
  int64_t n = mode <= 1 ? 8192 : 4096;
  Tensor out = at::empty({n}, at::TensorOptions()
                                  .dtype(at::kFloat)
                                  .device(at::kCUDA));
  const at::hip::OptionalHIPGuardMasqueradingAsCUDA guard(out.device());
  tr16_diag_launch(out.data_ptr<float>(), (int)mode, current_stream());
  return out;
}

Tensor mfma_probe(Tensor A, Tensor B) {
  TORCH_CHECK(A.is_cuda() && A.scalar_type() == at::kBFloat16 &&
              A.is_contiguous() && A.size(0) == 16 && A.size(1) == 32,
              "A must be [16,32] bf16");
  TORCH_CHECK(B.is_contiguous() && B.size(0) == 32 && B.size(1) == 16,
              "B must be [32,16] bf16");
  const at::hip::OptionalHIPGuardMasqueradingAsCUDA guard(A.device());
  Tensor D = at::empty({16, 16}, A.options().dtype(at::kFloat));
  mfma_probe_launch(A.data_ptr(), B.data_ptr(), D.data_ptr<float>(),
                    current_stream());
  return D;
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "machin_amd gfx950 HIP kernels";
  m.def("sumtree_update", &sumtree_update);
  m.def("sumtree_build", &sumtree_build);
  m.def("sumtree_sample", &sumtree_sample);
  m.def("discounted_returns", &discounted_returns);
  m.def("gae", &gae);
  m.def("vtrace", &vtrace);
  m.def("vtrace_bt", &vtrace_bt);
  m.def("categorical_projection", &categorical_projection);
  m.def("multi_tensor_polyak", &multi_tensor_polyak);
  m.def("multi_tensor_polyak_cached", &multi_tensor_polyak_cached);
  m.def("fused_rmsprop_cached", &fused_rmsprop_cached);
  m.def("nstep_returns", &nstep_returns);
  m.def("gaussian_sample_logprob", &gaussian_sample_logprob);
  m.def("gaussian_logprob", &gaussian_logprob);
  m.def("normal_noise_", &normal_noise_);
  m.def("ou_update_", &ou_update_);
  m.def("u8_to_bf16_scale", &u8_to_bf16_scale);
  m.def("pg_head_fwd", &pg_head_fwd);
  m.def("pg_head_bwd", &pg_head_bwd);
  m.def("conv1_wrw", &conv1_wrw);
  m.def("mfma_probe", &mfma_probe);
  m.def("tr16_probe", &tr16_probe);
  m.def("fwd_bfrag_probe", &fwd_bfrag_probe);
  m.def("tr16_diag", &tr16_diag);
  m.def("conv1_fwd", &conv1_fwd);
}
