// pybind bindings for the rl_amd native extension (`rl_amd._C`).
//
// Mirrors the reference's pybind module `torchrl._torchrl`
// (pytorch/rl torchrl/csrc/pybind.cpp:21-38): CPU segment trees +
// device kernels + safetanh — here the device side is HIP/CDNA4 and the
// value scans are first-class.

#include <torch/extension.h>

#include <vector>

#include "segment_tree_cpu.h"

#ifdef RL_AMD_WITH_HIP
#include <ATen/hip/HIPContext.h>
#include <c10/hip/HIPStream.h>

extern "C" {
void launch_gae_f32(const float*, const float*, const float*, const bool*,
                    const bool*, float*, float*, long, long, float, float,
                    void*);
void launch_gae_bf16(const void*, const void*, const void*, const bool*,
                     const bool*, void*, void*, long, long, float, float,
                     void*);
void launch_revscan_f32(const float*, const float*, float*, long, long, void*);
void launch_revscan_bf16(const void*, const void*, void*, long, long, void*);
void launch_vtrace_f32(const float*, const float*, const float*, const float*,
                       const float*, const bool*, const bool*, float*, float*,
                       long, long, float, float, float, void*);
void launch_tree_scan_f64(const double*, const double*, long*, long, long,
                          long, void*);
void launch_tree_update_f64(double*, double*, int*, const long*, const double*,
                            long, long, int, void*);
int gru_fused_lds_bytes(int);
int fused_actor_lds_bytes(int, int, int, int);
void launch_fused_actor(const float*, const float*, const float*, const float*,
                        const float*, const float*, const float*, const float*,
                        float*, float*, float*, float*, long, long, int, int,
                        int, int, int, float, float, void*);
int lstm_fused_lds_bytes(int);
int wgrad_slab_count(long);
int mlp3_lds_bytes(int, int, int);
int fused_rollout_lds_bytes(int, int, int, int);
int fused_rollout_mfma_ok(int, int, int, int);
void launch_fused_rollout_mfma(float*, float*, const void*, const void*,
                               const void*, const void*, const void*,
                               const void*, const float*, const float*,
                               const float*, const float*, float*, float*,
                               float*, float*, float*, bool*, int, int, int,
                               int, int, int, float, float, float, void*);
void launch_fused_rollout(float*, float*, const float*, const float*,
                          const float*, const float*, const float*,
                          const float*, const float*, const float*,
                          const float*, const float*, float*, float*, float*,
                          float*, float*, bool*, int, int, int, int, int, int,
                          float, float, float, void*);
void launch_mlp3_fwd(const void*, const void*, const void*, const void*,
                     const void*, const void*, const void*, void*, void*,
                     void*, int, int, int, int, void*);
void launch_mlp3_bwd(const void*, const void*, const void*, const void*,
                     const void*, void*, void*, int, int, int, void*);
int synthetic_env_step_lds_bytes(int, int);
void launch_tanh_normal_logprob_fwd(const float*, const float*, const float*,
                                    float*, int, int, void*);
void launch_tanh_normal_logprob_bwd(const float*, const float*, const float*,
                                    const float*, float*, float*, int, int,
                                    void*);
void launch_tanh_normal_entropy_fwd(const float*, const float*, const float*,
                                    float*, int, int, void*);
void launch_tanh_normal_entropy_bwd(const float*, const float*, const float*,
                                    const float*, float*, float*, int, int,
                                    void*);
void launch_adv_stats(const float*, float*, float*, long, void*);
void launch_adv_stats_batch(const float*, float*, float*, long, int, void*);
void launch_ppo_clip_fwd(const float*, const float*, const float*, float*,
                         float*, float*, float*, float, float, long, void*);
void launch_ppo_clip_bwd(const float*, const float*, const float*,
                         const float*, float*, float, float, long, void*);
void launch_smooth_l1_fwd(const void*, const float*, float*, float*, float,
                          long, int, void*);
void launch_smooth_l1_bwd(const void*, const float*, const float*, void*,
                          float, long, int, void*);
void launch_ppo_head_fwd(const void*, const float*, const float*,
                         const float*, const float*, const float*,
                         const void*, const float*, float*, float* const*,
                         float, float, float, float, float, float, long, int,
                         int, void*);
void launch_ppo_head_bwd(const void*, const float*, const float*,
                         const float*, const float*, const float*,
                         const void*, const float*, const float*,
                         const float*, const float*, const float*,
                         const float*, void*, void*, float, float, float,
                         float, float, float, long, int, int, void*);
int mlp3_mfma_lds_bytes(int, int, int);
int wgrad3_slab_count(long);
int wgrad3_slab_count_n(long, int);
void launch_grad_clip_coef(const void*, int, float, float*, float*, int,
                           void*);
void launch_multi_gather(const void*, int, const long*, long, void*);
void launch_multi_shuffle(const void*, int, const int*, long, void*);
void launch_wgrad3(const void* const*, const void* const*, float* const*,
                   float* const*, float* const*, float* const*, const int*,
                   const int*, int, long, float*, void*);
void launch_wgrad_clip_finalize(const float*, int, float, float*, int,
                                void*);
void launch_mlp3_mfma_fwd2(const void*, int, const void* const*,
                           void* const*, void*, long, int, const int*,
                           const int*, void*);
void launch_mlp3_mfma_bwd2(const void* const*, void* const*, long,
                           const int*, const int*, void*);
void launch_mlp3_mfma_fwdpair(const void*, const void*, int,
                              const void* const*, void*, void*, void*, long,
                              int, int, int, void*);
void launch_mlp3_mfma_fwd2_loss(const void*, int, const void* const*,
                                void* const*, void*, const float*,
                                const float*, const float*, const float*,
                                const float*, const float*, float*,
                                float* const*, float, float, float, float,
                                float, float, long, int, const int*,
                                const int*, int, void*);
void launch_mlp3_mfma_bwd2_loss(const void* const*, void* const*,
                                const float*, const float*, const float*,
                                const float*, const float*, const float*,
                                const float*, const float*, const float*,
                                const float*, const float*, void*, void*,
                                const void*, const void*, float, float,
                                float, float, float, float, long,
                                const int*, const int*, int, void*);
void launch_mlp3_mfma_fwd(const void*, int, const void*, const void*,
                          const void*, const void*, const void*, const void*,
                          void*, void*, void*, void*, long, int, int, int,
                          void*);
void launch_mlp3_mfma_bwd(const void*, int, const void*, const void*,
                          const void*, const void*, void*, void*, long, int,
                          int, void*);
void launch_synthetic_env_step(float*, const float*, const float*,
                               const float*, float*, float*, float*, float*,
                               bool*, const float*, long, long, long, int,
                               int, int, float, void*);
void launch_wgrad_splitk(const void*, const void*, float*, float*, float*,
                         float*, long, int, int, void*);
void launch_gru_fused(const float*, const float*, const float*, const bool*,
                      const float*, float*, float*, int, int, int, void*);
void launch_lstm_fused(const float*, const float*, const bool*, const float*,
                       const float*, float*, float*, float*, int, int, int,
                       void*);
void launch_gru_train_fwd(const float*, const void*, const float*,
                          const bool*, const float*, float*, float*, int,
                          int, int, void*);
void launch_gru_bwd(const float*, const void*, const void*, const float*,
                    const bool*, const float*, const float*, const float*,
                    float*, float*, float*, float*, float*, int, int, int,
                    void*);
void launch_lstm_train_fwd(const float*, const void*, const bool*,
                           const float*, const float*, float*, float*,
                           float*, int, int, int, void*);
void launch_lstm_bwd(const float*, const void*, const void*, const bool*,
                     const float*, const float*, const float*, const float*,
                     const float*, float*, float*, float*, float*, float*,
                     int, int, int, void*);
}

static void check_gae_args(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on a HIP device");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

// rows = product of batch dims; tensors viewed [rows, T]
std::vector<torch::Tensor> gae(torch::Tensor reward, torch::Tensor value,
                               torch::Tensor next_value, torch::Tensor done,
                               torch::Tensor terminated, double gamma,
                               double lmbda) {
  check_gae_args(reward, "reward");
  auto sizes = reward.sizes();
  long T = sizes[sizes.size() - 1];
  long B = reward.numel() / T;
  auto adv = torch::empty_like(reward);
  auto vt = torch::empty_like(reward);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  if (reward.scalar_type() == torch::kFloat32) {
    launch_gae_f32(reward.data_ptr<float>(), value.data_ptr<float>(),
                   next_value.data_ptr<float>(), done.data_ptr<bool>(),
                   terminated.data_ptr<bool>(), adv.data_ptr<float>(),
                   vt.data_ptr<float>(), B, T, (float)gamma, (float)lmbda,
                   (void*)stream);
  } else if (reward.scalar_type() == torch::kBFloat16) {
    launch_gae_bf16(reward.data_ptr(), value.data_ptr(), next_value.data_ptr(),
                    done.data_ptr<bool>(), terminated.data_ptr<bool>(),
                    adv.data_ptr(), vt.data_ptr(), B, T, (float)gamma,
                    (float)lmbda, (void*)stream);
  } else {
    TORCH_CHECK(false, "gae: dtype must be float32 or bfloat16");
  }
  return {adv, vt};
}

torch::Tensor revscan(torch::Tensor a, torch::Tensor b) {
  check_gae_args(a, "a");
  auto sizes = a.sizes();
  long T = sizes[sizes.size() - 1];
  long B = a.numel() / T;
  auto y = torch::empty_like(b);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  if (a.scalar_type() == torch::kFloat32) {
    launch_revscan_f32(a.data_ptr<float>(), b.data_ptr<float>(),
                       y.data_ptr<float>(), B, T, (void*)stream);
  } else if (a.scalar_type() == torch::kBFloat16) {
    launch_revscan_bf16(a.data_ptr(), b.data_ptr(), y.data_ptr(), B, T,
                        (void*)stream);
  } else {
    TORCH_CHECK(false, "revscan: dtype must be float32 or bfloat16");
  }
  return y;
}

std::vector<torch::Tensor> vtrace(torch::Tensor log_pi, torch::Tensor log_mu,
                                  torch::Tensor reward, torch::Tensor value,
                                  torch::Tensor next_value, torch::Tensor done,
                                  torch::Tensor terminated, double gamma,
                                  double rho_thresh, double c_thresh) {
  check_gae_args(reward, "reward");
  TORCH_CHECK(reward.scalar_type() == torch::kFloat32,
              "vtrace: fp32 only (cast inputs)");
  auto sizes = reward.sizes();
  long T = sizes[sizes.size() - 1];
  long B = reward.numel() / T;
  auto adv = torch::empty_like(reward);
  auto vs = torch::empty_like(reward);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  launch_vtrace_f32(log_pi.data_ptr<float>(), log_mu.data_ptr<float>(),
                    reward.data_ptr<float>(), value.data_ptr<float>(),
                    next_value.data_ptr<float>(), done.data_ptr<bool>(),
                    terminated.data_ptr<bool>(), adv.data_ptr<float>(),
                    vs.data_ptr<float>(), B, T, (float)gamma, (float)rho_thresh,
                    (float)c_thresh, (void*)stream);
  return {adv, vs};
}

// Device segment trees: fused inverse-CDF descent over a sum tree in HBM.
torch::Tensor tree_scan_lower_bound(torch::Tensor tree, torch::Tensor mass,
                                    long size, long capacity) {
  TORCH_CHECK(tree.is_cuda() && mass.is_cuda(), "tree/mass must be on device");
  TORCH_CHECK(tree.scalar_type() == torch::kFloat64, "tree must be f64");
  auto out = torch::empty({mass.numel()},
                          torch::dtype(torch::kLong).device(mass.device()));
  auto stream = c10::hip::getCurrentHIPStream().stream();
  launch_tree_scan_f64(tree.data_ptr<double>(), mass.data_ptr<double>(),
                       out.data_ptr<long>(), mass.numel(), size, capacity,
                       (void*)stream);
  return out;
}

// Fused leaf-scatter + two-pass path recompute; `cnt` is a persistent
// int32 workspace of 2*size zeros (self-restoring across calls).
void tree_update(torch::Tensor sum_tree, torch::Tensor min_tree,
                 torch::Tensor cnt, torch::Tensor index, torch::Tensor value,
                 long size, bool with_min) {
  TORCH_CHECK(sum_tree.is_cuda(), "tree must be on device");
  TORCH_CHECK(cnt.scalar_type() == torch::kInt32, "cnt workspace must be i32");
  auto stream = c10::hip::getCurrentHIPStream().stream();
  launch_tree_update_f64(sum_tree.data_ptr<double>(),
                         with_min ? min_tree.data_ptr<double>() : nullptr,
                         cnt.data_ptr<int>(), index.data_ptr<long>(),
                         value.data_ptr<double>(), index.numel(), size,
                         with_min ? 1 : 0, (void*)stream);
}
// Fused GRU forward scan: gates_x = x@W_ih + b_ih precomputed outside
// (one hipBLASLt GEMM over B*T); returns (ys, h_final).
std::vector<torch::Tensor> gru_fused(torch::Tensor gates_x, torch::Tensor w_hh,
                                     torch::Tensor bias_hh,
                                     torch::Tensor is_init,
                                     torch::Tensor h0) {
  TORCH_CHECK(gates_x.is_cuda() && gates_x.is_contiguous());
  long B = gates_x.size(0), T = gates_x.size(1);
  long H = gates_x.size(2) / 3;
  TORCH_CHECK(gru_fused_lds_bytes(H) <= 160 * 1024,
              "H too large for the fused GRU LDS budget");
  auto ys = torch::empty({B, T, H}, gates_x.options());
  auto h_out = torch::empty({B, H}, gates_x.options());
  auto stream = c10::hip::getCurrentHIPStream().stream();
  launch_gru_fused(gates_x.data_ptr<float>(), w_hh.data_ptr<float>(),
                   bias_hh.data_ptr<float>(), is_init.data_ptr<bool>(),
                   h0.numel() ? h0.data_ptr<float>() : nullptr,
                   ys.data_ptr<float>(), h_out.data_ptr<float>(), B, T, H,
                   (void*)stream);
  return {ys, h_out};
}

std::vector<torch::Tensor> lstm_fused(torch::Tensor gates_x,
                                      torch::Tensor w_hh,
                                      torch::Tensor is_init, torch::Tensor h0,
                                      torch::Tensor c0) {
  TORCH_CHECK(gates_x.is_cuda() && gates_x.is_contiguous());
  long B = gates_x.size(0), T = gates_x.size(1);
  long H = gates_x.size(2) / 4;
  TORCH_CHECK(lstm_fused_lds_bytes(H) <= 160 * 1024,
              "H too large for the fused LSTM LDS budget");
  auto ys = torch::empty({B, T, H}, gates_x.options());
  auto h_out = torch::empty({B, H}, gates_x.options());
  auto c_out = torch::empty({B, H}, gates_x.options());
  auto stream = c10::hip::getCurrentHIPStream().stream();
  launch_lstm_fused(gates_x.data_ptr<float>(), w_hh.data_ptr<float>(),
                    is_init.data_ptr<bool>(),
                    h0.numel() ? h0.data_ptr<float>() : nullptr,
                    c0.numel() ? c0.data_ptr<float>() : nullptr,
                    ys.data_ptr<float>(), h_out.data_ptr<float>(),
                    c_out.data_ptr<float>(), B, T, H, (void*)stream);
  return {ys, h_out, c_out};
}

// Training-path scans (reverse-time backward with gate recompute).
// ``wt`` is the pre-transposed bf16 W_hh^T [H, G*H]: staged to LDS when
// it fits the 160 KB budget, streamed from L2 otherwise (large H).
static const void* bf16_ptr(const torch::Tensor& t) {
  return (const void*)t.data_ptr<at::BFloat16>();
}

torch::Tensor gru_train_fwd(torch::Tensor gates_x, torch::Tensor wt,
                            torch::Tensor bias_hh, torch::Tensor is_init,
                            torch::Tensor h0) {
  TORCH_CHECK(gates_x.is_cuda() && gates_x.is_contiguous());
  TORCH_CHECK(wt.is_contiguous() && wt.scalar_type() == torch::kBFloat16,
              "wt must be contiguous bf16 [H, 3H]");
  long B = gates_x.size(0), T = gates_x.size(1);
  long H = gates_x.size(2) / 3;
  auto ys = torch::empty({B, T, H}, gates_x.options());
  long B_pad = (B + 15) / 16 * 16;
  auto gscr = torch::empty({B_pad * 3 * H}, gates_x.options());
  auto stream = c10::hip::getCurrentHIPStream().stream();
  launch_gru_train_fwd(gates_x.data_ptr<float>(), bf16_ptr(wt),
                       bias_hh.data_ptr<float>(), is_init.data_ptr<bool>(),
                       h0.numel() ? h0.data_ptr<float>() : nullptr,
                       ys.data_ptr<float>(), gscr.data_ptr<float>(), B, T, H,
                       (void*)stream);
  return ys;
}

std::vector<torch::Tensor> gru_bwd(torch::Tensor gates_x, torch::Tensor wt,
                                   torch::Tensor w_row,
                                   torch::Tensor bias_hh,
                                   torch::Tensor is_init, torch::Tensor h0,
                                   torch::Tensor ys, torch::Tensor dys) {
  TORCH_CHECK(gates_x.is_cuda() && gates_x.is_contiguous());
  TORCH_CHECK(dys.is_contiguous(), "dys must be contiguous");
  long B = gates_x.size(0), T = gates_x.size(1);
  long H = gates_x.size(2) / 3;
  auto dgx = torch::empty_like(gates_x);
  auto dgh = torch::empty_like(gates_x);
  auto hprev = torch::empty({B, T, H}, gates_x.options());
  auto dh0 = torch::empty({B, H}, gates_x.options());
  long B_pad = (B + 15) / 16 * 16;
  auto gscr = torch::empty({B_pad * 3 * H}, gates_x.options());
  auto stream = c10::hip::getCurrentHIPStream().stream();
  launch_gru_bwd(gates_x.data_ptr<float>(), bf16_ptr(wt), bf16_ptr(w_row),
                 bias_hh.data_ptr<float>(), is_init.data_ptr<bool>(),
                 h0.numel() ? h0.data_ptr<float>() : nullptr,
                 ys.data_ptr<float>(), dys.data_ptr<float>(),
                 dgx.data_ptr<float>(), dgh.data_ptr<float>(),
                 hprev.data_ptr<float>(), dh0.data_ptr<float>(),
                 gscr.data_ptr<float>(), B, T, H, (void*)stream);
  return {dgx, dgh, hprev, dh0};
}

std::vector<torch::Tensor> lstm_train_fwd(torch::Tensor gates_x,
                                          torch::Tensor wt,
                                          torch::Tensor is_init,
                                          torch::Tensor h0, torch::Tensor c0) {
  TORCH_CHECK(gates_x.is_cuda() && gates_x.is_contiguous());
  TORCH_CHECK(wt.is_contiguous() && wt.scalar_type() == torch::kBFloat16,
              "wt must be contiguous bf16 [H, 4H]");
  long B = gates_x.size(0), T = gates_x.size(1);
  long H = gates_x.size(2) / 4;
  auto ys = torch::empty({B, T, H}, gates_x.options());
  auto cs = torch::empty({B, T, H}, gates_x.options());
  long B_pad = (B + 15) / 16 * 16;
  auto gscr = torch::empty({B_pad * 4 * H}, gates_x.options());
  auto stream = c10::hip::getCurrentHIPStream().stream();
  launch_lstm_train_fwd(gates_x.data_ptr<float>(), bf16_ptr(wt),
                        is_init.data_ptr<bool>(),
                        h0.numel() ? h0.data_ptr<float>() : nullptr,
                        c0.numel() ? c0.data_ptr<float>() : nullptr,
                        ys.data_ptr<float>(), cs.data_ptr<float>(),
                        gscr.data_ptr<float>(), B, T, H, (void*)stream);
  return {ys, cs};
}

std::vector<torch::Tensor> lstm_bwd(torch::Tensor gates_x, torch::Tensor wt,
                                    torch::Tensor w_row,
                                    torch::Tensor is_init, torch::Tensor h0,
                                    torch::Tensor c0, torch::Tensor ys,
                                    torch::Tensor cs, torch::Tensor dys) {
  TORCH_CHECK(gates_x.is_cuda() && gates_x.is_contiguous());
  TORCH_CHECK(dys.is_contiguous(), "dys must be contiguous");
  long B = gates_x.size(0), T = gates_x.size(1);
  long H = gates_x.size(2) / 4;
  auto dg = torch::empty_like(gates_x);
  auto hprev = torch::empty({B, T, H}, gates_x.options());
  auto dh0 = torch::empty({B, H}, gates_x.options());
  auto dc0 = torch::empty({B, H}, gates_x.options());
  long B_pad = (B + 15) / 16 * 16;
  auto gscr = torch::empty({B_pad * 4 * H}, gates_x.options());
  auto stream = c10::hip::getCurrentHIPStream().stream();
  launch_lstm_bwd(gates_x.data_ptr<float>(), bf16_ptr(wt), bf16_ptr(w_row),
                  is_init.data_ptr<bool>(),
                  h0.numel() ? h0.data_ptr<float>() : nullptr,
                  c0.numel() ? c0.data_ptr<float>() : nullptr,
                  ys.data_ptr<float>(), cs.data_ptr<float>(),
                  dys.data_ptr<float>(), dg.data_ptr<float>(),
                  hprev.data_ptr<float>(), dh0.data_ptr<float>(),
                  dc0.data_ptr<float>(), gscr.data_ptr<float>(), B, T, H,
                  (void*)stream);
  return {dg, hprev, dh0, dc0};
}

// Fused actor: MLP(tanh)x2 + heads + TanhNormal sample + log-prob.
std::vector<torch::Tensor> fused_actor(
    torch::Tensor obs, torch::Tensor w1, torch::Tensor b1, torch::Tensor w2,
    torch::Tensor b2, torch::Tensor w3, torch::Tensor b3, torch::Tensor eps,
    double inv_softplus_bias, double scale_lb, bool want_loc_scale) {
  TORCH_CHECK(obs.is_cuda() && obs.is_contiguous(), "obs must be device+contig");
  long B = obs.size(0), O = obs.size(1);
  long H1 = w1.size(0), H2 = w2.size(0), A = w3.size(0) / 2;
  TORCH_CHECK(fused_actor_lds_bytes(O, H1, H2, A) <= 160 * 1024,
              "fused actor exceeds the LDS budget; fall back to eager");
  auto action = torch::empty({B, A}, obs.options());
  auto logp = torch::empty({B}, obs.options());
  torch::Tensor loc, scale;
  if (want_loc_scale) {
    loc = torch::empty({B, A}, obs.options());
    scale = torch::empty({B, A}, obs.options());
  }
  auto stream = c10::hip::getCurrentHIPStream().stream();
  launch_fused_actor(
      obs.data_ptr<float>(), w1.data_ptr<float>(), b1.data_ptr<float>(),
      w2.data_ptr<float>(), b2.data_ptr<float>(), w3.data_ptr<float>(),
      b3.data_ptr<float>(), eps.data_ptr<float>(), action.data_ptr<float>(),
      logp.data_ptr<float>(),
      want_loc_scale ? loc.data_ptr<float>() : nullptr,
      want_loc_scale ? scale.data_ptr<float>() : nullptr, (long)A, 1L, B, O,
      H1, H2, A, (float)inv_softplus_bias, (float)scale_lb, (void*)stream);
  if (want_loc_scale) return {action, logp, loc, scale};
  return {action, logp};
}
// Split-K weight gradient: dW = dY^T X (+ dBias), bf16 in / fp32 accum.
std::vector<torch::Tensor> wgrad_splitk(torch::Tensor dy, torch::Tensor x,
                                        bool want_bias) {
  TORCH_CHECK(dy.is_cuda() && x.is_cuda(), "wgrad_splitk: device tensors");
  TORCH_CHECK(dy.scalar_type() == torch::kBFloat16 &&
                  x.scalar_type() == torch::kBFloat16,
              "wgrad_splitk: bf16 inputs");
  TORCH_CHECK(dy.is_contiguous() && x.is_contiguous(), "contiguous inputs");
  TORCH_CHECK(dy.size(0) == x.size(0), "K mismatch");
  long K = dy.size(0);
  int N = (int)dy.size(1), M = (int)x.size(1);
  auto opts = dy.options().dtype(torch::kFloat32);
  auto dw = torch::empty({N, M}, opts);  // wave-reduce writes every element
  torch::Tensor db;
  if (want_bias) db = torch::empty({N}, opts);
  const long slabs = wgrad_slab_count(K);
  const long tiles_n = (N + 63) / 64, tiles_m = (M + 63) / 64;
  auto part = torch::empty({slabs, tiles_n * tiles_m * 64 * 64}, opts);
  auto bias_part = torch::empty({slabs, tiles_n * 64}, opts);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  launch_wgrad_splitk(dy.data_ptr(), x.data_ptr(), dw.data_ptr<float>(),
                      want_bias ? db.data_ptr<float>() : nullptr,
                      part.data_ptr<float>(), bias_part.data_ptr<float>(), K,
                      N, M, (void*)stream);
  if (want_bias) return {dw, db};
  return {dw};
}
// Fused synthetic-env transition: one launch replaces the ~12-kernel
// eager step (GEMMs + tanh + reward + time/done bookkeeping).
std::vector<torch::Tensor> synthetic_env_step(torch::Tensor state,
                                              torch::Tensor action,
                                              torch::Tensor A, torch::Tensor B,
                                              torch::Tensor t,
                                              double max_steps) {
  TORCH_CHECK(state.is_cuda() && state.scalar_type() == torch::kFloat32,
              "synthetic_env_step: fp32 cuda state");
  TORCH_CHECK(state.is_contiguous() && action.is_contiguous() &&
                  A.is_contiguous() && B.is_contiguous() && t.is_contiguous(),
              "contiguous inputs");
  const int Bn = (int)state.size(0), S = (int)state.size(1),
            Aact = (int)action.size(1);
  TORCH_CHECK(synthetic_env_step_lds_bytes(S, Aact) <= 160 * 1024,
              "env dims exceed the LDS budget");
  auto obs = torch::empty_like(state);
  auto reward = torch::empty({Bn, 1}, state.options());
  auto done = torch::empty({Bn, 1}, state.options().dtype(torch::kBool));
  auto stream = c10::hip::getCurrentHIPStream().stream();
  launch_synthetic_env_step(state.data_ptr<float>(), action.data_ptr<float>(),
                            A.data_ptr<float>(), B.data_ptr<float>(),
                            t.data_ptr<float>(), obs.data_ptr<float>(),
                            nullptr, reward.data_ptr<float>(),
                            done.data_ptr<bool>(), nullptr, (long)S, 1,
                            (long)Aact, Bn, S, Aact, (float)max_steps,
                            (void*)stream);
  return {obs, reward, done};
}

// Store-direct variant: writes pre/post observations, reward and done
// straight into strided [B, T] rollout-store views and auto-resets the
// carried state from `reset_noise` — the 4-launch rollout step.
void synthetic_env_step_into(torch::Tensor state, torch::Tensor action,
                             torch::Tensor A, torch::Tensor B,
                             torch::Tensor t, torch::Tensor next_obs,
                             torch::Tensor prev_obs, torch::Tensor reward,
                             torch::Tensor done, torch::Tensor reset_noise,
                             double max_steps) {
  TORCH_CHECK(state.is_cuda() && state.scalar_type() == torch::kFloat32,
              "synthetic_env_step_into: fp32 cuda state");
  TORCH_CHECK(next_obs.stride(-1) == 1 && prev_obs.stride(-1) == 1,
              "obs views must be inner-contiguous");
  const int Bn = (int)state.size(0), S = (int)state.size(1),
            Aact = (int)action.size(-1);
  TORCH_CHECK(action.stride(-1) == 1, "action inner-contiguous");
  auto stream = c10::hip::getCurrentHIPStream().stream();
  launch_synthetic_env_step(
      state.data_ptr<float>(), action.data_ptr<float>(), A.data_ptr<float>(),
      B.data_ptr<float>(), t.data_ptr<float>(), next_obs.data_ptr<float>(),
      prev_obs.data_ptr<float>(), reward.data_ptr<float>(),
      done.data_ptr<bool>(), reset_noise.data_ptr<float>(),
      (long)next_obs.stride(0), (long)reward.stride(0),
      (long)action.stride(0), Bn, S, Aact, (float)max_steps, (void*)stream);
}
// Whole-rollout mega-kernel: the entire T-step rollout in one launch.
void fused_rollout(torch::Tensor state, torch::Tensor step_ct,
                   torch::Tensor w1, torch::Tensor b1, torch::Tensor w2,
                   torch::Tensor b2, torch::Tensor w3, torch::Tensor b3,
                   torch::Tensor A, torch::Tensor B, torch::Tensor eps,
                   torch::Tensor noise, torch::Tensor st_obs,
                   torch::Tensor st_act, torch::Tensor st_logp,
                   torch::Tensor st_nobs, torch::Tensor st_rew,
                   torch::Tensor st_done, double max_steps,
                   double inv_softplus_bias, double scale_lb,
                   std::vector<torch::Tensor> bf16_weights) {
  TORCH_CHECK(state.is_cuda() && state.scalar_type() == torch::kFloat32,
              "fused_rollout: fp32 cuda state");
  const int Bn = (int)state.size(0), S = (int)state.size(1);
  const int H1 = (int)w1.size(0), H2 = (int)w2.size(0);
  const int Aact = (int)w3.size(0) / 2;
  const int T = (int)eps.size(0);
  TORCH_CHECK(fused_rollout_lds_bytes(S, H1, H2, Aact) <= 160 * 1024,
              "fused_rollout exceeds the LDS budget");
  TORCH_CHECK(st_obs.is_contiguous() && st_act.is_contiguous() &&
                  st_logp.is_contiguous() && st_nobs.is_contiguous() &&
                  st_rew.is_contiguous() && st_done.is_contiguous(),
              "store tensors must be contiguous [B, T, ...]");
  auto stream = c10::hip::getCurrentHIPStream().stream();
  if (bf16_weights.size() == 6 &&
      fused_rollout_mfma_ok(S, H1, H2, Aact)) {
    for (auto& t : bf16_weights)
      TORCH_CHECK(t.scalar_type() == torch::kBFloat16 && t.is_contiguous(),
                  "fused_rollout: bf16 cache tensors");
    launch_fused_rollout_mfma(
        state.data_ptr<float>(), step_ct.data_ptr<float>(),
        bf16_weights[0].data_ptr(), bf16_weights[1].data_ptr(),
        bf16_weights[2].data_ptr(), bf16_weights[3].data_ptr(),
        bf16_weights[4].data_ptr(), bf16_weights[5].data_ptr(),
        A.data_ptr<float>(), B.data_ptr<float>(), eps.data_ptr<float>(),
        noise.data_ptr<float>(), st_obs.data_ptr<float>(),
        st_act.data_ptr<float>(), st_logp.data_ptr<float>(),
        st_nobs.data_ptr<float>(), st_rew.data_ptr<float>(),
        st_done.data_ptr<bool>(), Bn, S, H1, H2, Aact, T, (float)max_steps,
        (float)inv_softplus_bias, (float)scale_lb, (void*)stream);
    return;
  }
  launch_fused_rollout(
      state.data_ptr<float>(), step_ct.data_ptr<float>(),
      w1.data_ptr<float>(), b1.data_ptr<float>(), w2.data_ptr<float>(),
      b2.data_ptr<float>(), w3.data_ptr<float>(), b3.data_ptr<float>(),
      A.data_ptr<float>(), B.data_ptr<float>(), eps.data_ptr<float>(),
      noise.data_ptr<float>(), st_obs.data_ptr<float>(),
      st_act.data_ptr<float>(), st_logp.data_ptr<float>(),
      st_nobs.data_ptr<float>(), st_rew.data_ptr<float>(),
      st_done.data_ptr<bool>(), Bn, S, H1, H2, Aact, T, (float)max_steps,
      (float)inv_softplus_bias, (float)scale_lb, (void*)stream);
}

// Fused 3-layer MLP forward/backward-dgrad (PPO update phase).
std::vector<torch::Tensor> mlp3_fwd(torch::Tensor x, torch::Tensor w1,
                                    torch::Tensor b1, torch::Tensor w2,
                                    torch::Tensor b2, torch::Tensor w3,
                                    torch::Tensor b3) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16,
              "mlp3_fwd: bf16 cuda input");
  TORCH_CHECK(x.is_contiguous(), "contiguous input");
  const int N = (int)x.size(0), O = (int)x.size(1);
  const int H = (int)w1.size(0), A2 = (int)w3.size(0);
  TORCH_CHECK(mlp3_lds_bytes(O, H, A2) <= 160 * 1024, "MLP exceeds LDS budget");
  auto out = torch::empty({N, A2}, x.options());
  auto h1 = torch::empty({N, H}, x.options());
  auto h2 = torch::empty({N, H}, x.options());
  auto stream = c10::hip::getCurrentHIPStream().stream();
  launch_mlp3_fwd(x.data_ptr(), w1.data_ptr(), b1.data_ptr(), w2.data_ptr(),
                  b2.data_ptr(), w3.data_ptr(), b3.data_ptr(), out.data_ptr(),
                  h1.data_ptr(), h2.data_ptr(), N, O, H, A2, (void*)stream);
  return {out, h1, h2};
}

std::vector<torch::Tensor> mlp3_bwd(torch::Tensor dout, torch::Tensor h1,
                                    torch::Tensor h2, torch::Tensor w2,
                                    torch::Tensor w3) {
  const int N = (int)dout.size(0), A2 = (int)dout.size(1);
  const int H = (int)h1.size(1);
  auto dh1 = torch::empty_like(h1);
  auto dh2 = torch::empty_like(h2);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  launch_mlp3_bwd(dout.contiguous().data_ptr(), h1.data_ptr(), h2.data_ptr(),
                  w2.data_ptr(), w3.data_ptr(), dh1.data_ptr(), dh2.data_ptr(),
                  N, H, A2, (void*)stream);
  return {dh1, dh2};
}

// Fused TanhNormal log-prob (PPO ratio; action is data).
torch::Tensor tanh_normal_logprob(torch::Tensor loc, torch::Tensor scale,
                                  torch::Tensor action) {
  TORCH_CHECK(loc.is_cuda() && loc.scalar_type() == torch::kFloat32,
              "tanh_normal_logprob: fp32 cuda");
  TORCH_CHECK(loc.is_contiguous() && scale.is_contiguous() &&
                  action.is_contiguous(),
              "contiguous inputs");
  const int N = (int)loc.size(0), A = (int)loc.size(1);
  auto logp = torch::empty({N}, loc.options());
  auto stream = c10::hip::getCurrentHIPStream().stream();
  launch_tanh_normal_logprob_fwd(loc.data_ptr<float>(),
                                 scale.data_ptr<float>(),
                                 action.data_ptr<float>(),
                                 logp.data_ptr<float>(), N, A, (void*)stream);
  return logp;
}

std::vector<torch::Tensor> tanh_normal_logprob_bwd(torch::Tensor loc,
                                                   torch::Tensor scale,
                                                   torch::Tensor action,
                                                   torch::Tensor gout) {
  const int N = (int)loc.size(0), A = (int)loc.size(1);
  auto dloc = torch::empty_like(loc);
  auto dscale = torch::empty_like(scale);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  launch_tanh_normal_logprob_bwd(
      loc.data_ptr<float>(), scale.data_ptr<float>(), action.data_ptr<float>(),
      gout.contiguous().data_ptr<float>(), dloc.data_ptr<float>(),
      dscale.data_ptr<float>(), N, A, (void*)stream);
  return {dloc, dscale};
}

torch::Tensor tanh_normal_entropy(torch::Tensor loc, torch::Tensor scale,
                                  torch::Tensor eps) {
  const int N = (int)loc.size(0), A = (int)loc.size(1);
  auto ent = torch::empty({N}, loc.options());
  auto stream = c10::hip::getCurrentHIPStream().stream();
  launch_tanh_normal_entropy_fwd(loc.data_ptr<float>(),
                                 scale.data_ptr<float>(),
                                 eps.data_ptr<float>(), ent.data_ptr<float>(),
                                 N, A, (void*)stream);
  return ent;
}

std::vector<torch::Tensor> tanh_normal_entropy_bwd(torch::Tensor loc,
                                                   torch::Tensor scale,
                                                   torch::Tensor eps,
                                                   torch::Tensor gout) {
  const int N = (int)loc.size(0), A = (int)loc.size(1);
  auto dloc = torch::empty_like(loc);
  auto dscale = torch::empty_like(scale);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  launch_tanh_normal_entropy_bwd(
      loc.data_ptr<float>(), scale.data_ptr<float>(), eps.data_ptr<float>(),
      gout.contiguous().data_ptr<float>(), dloc.data_ptr<float>(),
      dscale.data_ptr<float>(), N, A, (void*)stream);
  return {dloc, dscale};
}

// Store-direct fused actor: action/log-prob written straight into
// strided [B, T] rollout-store views.
void fused_actor_into(torch::Tensor obs, torch::Tensor w1, torch::Tensor b1,
                      torch::Tensor w2, torch::Tensor b2, torch::Tensor w3,
                      torch::Tensor b3, torch::Tensor eps,
                      torch::Tensor action, torch::Tensor logp,
                      double inv_softplus_bias, double scale_lb) {
  TORCH_CHECK(obs.is_cuda() && obs.is_contiguous(), "obs device+contig");
  TORCH_CHECK(action.stride(-1) == 1, "action inner-contiguous");
  long B = obs.size(0), O = obs.size(1);
  long H1 = w1.size(0), H2 = w2.size(0), A = w3.size(0) / 2;
  auto stream = c10::hip::getCurrentHIPStream().stream();
  launch_fused_actor(
      obs.data_ptr<float>(), w1.data_ptr<float>(), b1.data_ptr<float>(),
      w2.data_ptr<float>(), b2.data_ptr<float>(), w3.data_ptr<float>(),
      b3.data_ptr<float>(), eps.data_ptr<float>(), action.data_ptr<float>(),
      logp.data_ptr<float>(), nullptr, nullptr, (long)action.stride(0),
      (long)logp.stride(0), B, O, H1, H2, A, (float)inv_softplus_bias,
      (float)scale_lb, (void*)stream);
}
// Fused ClipPPO objective (reference torchrl/objectives/ppo.py:1082
// ClipPPOLoss.forward elementwise/reduction chain).  Returns
// (out[3] = {loss_objective, ESS/N, clip_fraction}, stats[2]) where
// stats = (mu, 1/sigma) of the optional advantage normalization —
// saved for backward so the normalized advantage is never
// materialized.
std::vector<torch::Tensor> ppo_clip_fwd(torch::Tensor lw, torch::Tensor adv,
                                        double lo, double hi,
                                        bool normalize) {
  TORCH_CHECK(lw.is_cuda() && lw.scalar_type() == torch::kFloat32 &&
                  adv.scalar_type() == torch::kFloat32,
              "ppo_clip_fwd: fp32 cuda");
  TORCH_CHECK(lw.is_contiguous() && adv.is_contiguous(), "contiguous");
  const long N = lw.numel();
  TORCH_CHECK(adv.numel() == N, "ppo_clip_fwd: numel mismatch");
  auto opt = lw.options();
  auto part = torch::empty({256 * 4}, opt);
  auto o_loss = torch::empty({}, opt);
  auto o_ess = torch::empty({}, opt);
  auto o_cf = torch::empty({}, opt);
  auto stats = normalize ? torch::empty({2}, opt) : torch::empty({0}, opt);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  if (normalize)
    launch_adv_stats(adv.data_ptr<float>(), part.data_ptr<float>(),
                     stats.data_ptr<float>(), N, (void*)stream);
  launch_ppo_clip_fwd(lw.data_ptr<float>(), adv.data_ptr<float>(),
                      normalize ? stats.data_ptr<float>() : nullptr,
                      part.data_ptr<float>(), o_loss.data_ptr<float>(),
                      o_ess.data_ptr<float>(), o_cf.data_ptr<float>(),
                      (float)lo, (float)hi, N, (void*)stream);
  return {o_loss, o_ess, o_cf, stats};
}

torch::Tensor ppo_clip_bwd(torch::Tensor lw, torch::Tensor adv,
                           torch::Tensor stats, torch::Tensor gout,
                           double lo, double hi) {
  const long N = lw.numel();
  auto dlw = torch::empty_like(lw);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  launch_ppo_clip_bwd(
      lw.data_ptr<float>(), adv.data_ptr<float>(),
      stats.numel() ? stats.data_ptr<float>() : nullptr,
      gout.data_ptr<float>(), dlw.data_ptr<float>(), (float)lo, (float)hi, N,
      (void*)stream);
  return dlw;
}

// Fused mean smooth-L1 (beta=1) critic loss; value may be bf16 (autocast).
torch::Tensor smooth_l1_fwd(torch::Tensor v, torch::Tensor t,
                            double scale) {
  TORCH_CHECK(v.is_cuda() && v.is_contiguous() && t.is_contiguous(),
              "smooth_l1: cuda contiguous");
  TORCH_CHECK(t.scalar_type() == torch::kFloat32, "target fp32");
  const bool bf16 = v.scalar_type() == torch::kBFloat16;
  TORCH_CHECK(bf16 || v.scalar_type() == torch::kFloat32, "value fp32/bf16");
  const long N = v.numel();
  TORCH_CHECK(t.numel() == N, "smooth_l1: numel mismatch");
  auto part = torch::empty({256}, t.options());
  auto out = torch::empty({}, t.options());
  auto stream = c10::hip::getCurrentHIPStream().stream();
  launch_smooth_l1_fwd(v.data_ptr(), t.data_ptr<float>(),
                       part.data_ptr<float>(), out.data_ptr<float>(),
                       (float)scale, N, bf16 ? 1 : 0, (void*)stream);
  return out;
}

torch::Tensor smooth_l1_bwd(torch::Tensor v, torch::Tensor t,
                            torch::Tensor gout, double scale) {
  const bool bf16 = v.scalar_type() == torch::kBFloat16;
  const long N = v.numel();
  auto dv = torch::empty_like(v);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  launch_smooth_l1_bwd(v.data_ptr(), t.data_ptr<float>(),
                       gout.data_ptr<float>(), dv.data_ptr(), (float)scale,
                       N, bf16 ? 1 : 0, (void*)stream);
  return dv;
}
// Fused gradient clipping (csrc/loss_ops.hip): one single-workgroup
// kernel computes the global 2-norm over up to 32 fp32 gradients and
// the clamped scale coefficient (torch's clip_grad_norm_ runs ~10
// launches for the same).  Apply with torch._foreach_mul_(grads, coef).
torch::Tensor fused_grad_clip_coef(std::vector<torch::Tensor> grads,
                                   double max_norm, bool inverse,
                                   torch::Tensor out) {
  struct {
    const float* g[32];
    int len[32];
  } args;
  TORCH_CHECK(!grads.empty() && grads.size() <= 32, "1..32 gradients");
  for (size_t i = 0; i < grads.size(); ++i) {
    TORCH_CHECK(grads[i].is_cuda() &&
                    grads[i].scalar_type() == torch::kFloat32 &&
                    grads[i].is_contiguous(),
                "fused_grad_clip: fp32 cuda contiguous");
    args.g[i] = grads[i].data_ptr<float>();
    args.len[i] = (int)grads[i].numel();
  }
  auto coef = out.numel() ? out : torch::empty({}, grads[0].options());
  auto part = torch::empty({32}, grads[0].options());
  auto stream = c10::hip::getCurrentHIPStream().stream();
  launch_grad_clip_coef(&args, (int)grads.size(), (float)max_norm,
                        part.data_ptr<float>(), coef.data_ptr<float>(),
                        inverse ? 1 : 0, (void*)stream);
  return coef;
}

// Batched shuffle gather: dst[t][i] = src[t][perm[i]] for up to 8 fp32
// row-major tensors sharing the leading dim (csrc/loss_ops.hip) — one
// launch instead of one index kernel per tensordict key.
std::vector<torch::Tensor> multi_gather(torch::Tensor perm,
                                        std::vector<torch::Tensor> srcs) {
  struct {
    const float* src[8];
    float* dst[8];
    int w[8];
  } args;
  TORCH_CHECK(perm.is_cuda() && perm.scalar_type() == torch::kLong &&
                  perm.is_contiguous(),
              "perm int64 cuda");
  const long n = perm.numel();
  TORCH_CHECK(!srcs.empty() && srcs.size() <= 8, "1..8 tensors");
  std::vector<torch::Tensor> out;
  for (size_t i = 0; i < srcs.size(); ++i) {
    auto& s = srcs[i];
    TORCH_CHECK(s.is_cuda() && s.scalar_type() == torch::kFloat32 &&
                    s.is_contiguous() && s.size(0) == n,
                "multi_gather: fp32 cuda contiguous with shared dim 0");
    auto d = torch::empty_like(s);
    args.src[i] = s.data_ptr<float>();
    args.dst[i] = d.data_ptr<float>();
    args.w[i] = (int)(s.numel() / n);
    out.push_back(d);
  }
  auto stream = c10::hip::getCurrentHIPStream().stream();
  launch_multi_gather(&args, (int)srcs.size(), perm.data_ptr<long>(), n,
                      (void*)stream);
  return out;
}

// Feistel-shuffle variant: dst[t][i] = src[t][perm(i)] where perm is a
// keyed 4-round Feistel permutation computed inline — replaces
// randperm's radix sort + per-key index kernels with ONE launch.
std::vector<torch::Tensor> multi_shuffle(torch::Tensor keys,
                                         std::vector<torch::Tensor> srcs) {
  struct {
    const float* src[8];
    float* dst[8];
    int w[8];
  } args;
  TORCH_CHECK(keys.is_cuda() && keys.scalar_type() == torch::kInt32 &&
                  keys.numel() >= 4 && keys.is_contiguous(),
              "keys int32[4] cuda");
  TORCH_CHECK(!srcs.empty() && srcs.size() <= 8, "1..8 tensors");
  const long n = srcs[0].size(0);
  std::vector<torch::Tensor> out;
  for (size_t i = 0; i < srcs.size(); ++i) {
    auto& s = srcs[i];
    TORCH_CHECK(s.is_cuda() && s.scalar_type() == torch::kFloat32 &&
                    s.is_contiguous() && s.size(0) == n,
                "multi_shuffle: fp32 cuda contiguous with shared dim 0");
    auto d = torch::empty_like(s);
    args.src[i] = s.data_ptr<float>();
    args.dst[i] = d.data_ptr<float>();
    args.w[i] = (int)(s.numel() / n);
    out.push_back(d);
  }
  auto stream = c10::hip::getCurrentHIPStream().stream();
  launch_multi_shuffle(&args, (int)srcs.size(), keys.data_ptr<int>(), n,
                       (void*)stream);
  return out;
}

// Batched 3-layer wgrad (csrc/wgrad.hip): the three (dY, X) pairs of a
// fused-MLP backward in one mfma + one reduce launch.  All dims <= 64
// and a shared K are required (the PPO/critic MLP shape).
bool mlp3_mfma_ok(long O, long H, long A2);

std::vector<torch::Tensor> wgrad_splitk_batch_sq(
    std::vector<torch::Tensor> dys_v, std::vector<torch::Tensor> xs_v,
    torch::Tensor sq_part) {
  const int n_layers = (int)dys_v.size();
  TORCH_CHECK(n_layers >= 1 && n_layers <= 6 &&
                  xs_v.size() == dys_v.size(),
              "wgrad batch: 1..6 layer pairs");
  torch::Tensor* dys = dys_v.data();
  torch::Tensor* xs = xs_v.data();
  const long K = dys[0].size(0);
  const int slabs = wgrad3_slab_count_n(K, n_layers);
  const void* dyp[6];
  const void* xp[6];
  float *partp[6], *biasp[6], *dwp[6], *dbp[6];
  int N[6], M[6];
  std::vector<torch::Tensor> out;
  std::vector<torch::Tensor> keep;
  auto fopt = dys[0].options().dtype(torch::kFloat32);
  for (int l = 0; l < n_layers; ++l) {
    TORCH_CHECK(dys[l].scalar_type() == torch::kBFloat16 &&
                    xs[l].scalar_type() == torch::kBFloat16,
                "wgrad3: bf16 inputs");
    TORCH_CHECK(dys[l].is_contiguous() && xs[l].is_contiguous(),
                "wgrad3: contiguous");
    TORCH_CHECK(dys[l].size(0) == K && xs[l].size(0) == K, "wgrad3: K");
    N[l] = (int)dys[l].size(1);
    M[l] = (int)xs[l].size(1);
    TORCH_CHECK(N[l] <= 64 && M[l] <= 64, "wgrad3: dims <= 64");
    auto dw = torch::empty({N[l], M[l]}, fopt);
    auto db = torch::empty({N[l]}, fopt);
    auto part = torch::empty({slabs, 64 * 64}, fopt);
    auto bias_part = torch::empty({slabs, 64}, fopt);
    dyp[l] = dys[l].data_ptr();
    xp[l] = xs[l].data_ptr();
    partp[l] = part.data_ptr<float>();
    biasp[l] = bias_part.data_ptr<float>();
    dwp[l] = dw.data_ptr<float>();
    dbp[l] = db.data_ptr<float>();
    out.push_back(dw);
    out.push_back(db);
    keep.push_back(part);
    keep.push_back(bias_part);
  }
  auto stream = c10::hip::getCurrentHIPStream().stream();
  launch_wgrad3(dyp, xp, partp, biasp, dwp, dbp, N, M, n_layers, K,
                sq_part.numel() ? sq_part.data_ptr<float>() : nullptr,
                (void*)stream);
  return out;
}

std::vector<torch::Tensor> wgrad_splitk_batch(
    std::vector<torch::Tensor> dys_v, std::vector<torch::Tensor> xs_v) {
  return wgrad_splitk_batch_sq(dys_v, xs_v, torch::Tensor());
}

// grad-clip coefficient from the wgrad reduce's [n_layers*512] sq
// partials: call after wgrad_splitk_batch_sq covered EVERY gradient.
void wgrad_clip_finalize(torch::Tensor sq_part, double max_norm,
                         torch::Tensor coef, bool inverse) {
  auto stream = c10::hip::getCurrentHIPStream().stream();
  launch_wgrad_clip_finalize(sq_part.data_ptr<float>(),
                             (int)sq_part.numel(), (float)max_norm,
                             coef.data_ptr<float>(), inverse ? 1 : 0,
                             (void*)stream);
}

std::vector<torch::Tensor> wgrad_splitk3(torch::Tensor dy0, torch::Tensor x0,
                                         torch::Tensor dy1, torch::Tensor x1,
                                         torch::Tensor dy2,
                                         torch::Tensor x2) {
  return wgrad_splitk_batch({dy0, dy1, dy2}, {x0, x1, x2});
}

// Dual-network (actor+critic) whole-MLP forward/backward: both nets
// over the SAME input rows in one launch each way (grid.y = net).
std::vector<torch::Tensor> mlp3_mfma_fwd2(torch::Tensor x,
                                          std::vector<torch::Tensor> aw,
                                          std::vector<torch::Tensor> cw) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous(), "x cuda contiguous");
  TORCH_CHECK(aw.size() == 6 && cw.size() == 6, "6 weight tensors per net");
  const bool xf32 = x.scalar_type() == torch::kFloat32;
  const long N = x.size(0), O = x.size(1);
  int H[2] = {(int)aw[0].size(0), (int)cw[0].size(0)};
  int A2[2] = {(int)aw[4].size(0), (int)cw[4].size(0)};
  TORCH_CHECK(mlp3_mfma_ok(O, H[0], A2[0]) && mlp3_mfma_ok(O, H[1], A2[1]),
              "mlp3_mfma_fwd2: unsupported dims");
  auto bopt = aw[0].options();
  std::vector<torch::Tensor> res;
  const void* wp[12];
  void* op[6];
  for (int n = 0; n < 2; ++n) {
    auto& w = n == 0 ? aw : cw;
    for (int i = 0; i < 6; ++i) {
      TORCH_CHECK(w[i].scalar_type() == torch::kBFloat16 &&
                      w[i].is_contiguous(),
                  "bf16 contiguous weights");
      wp[n * 6 + i] = w[i].data_ptr();
    }
    auto outn = torch::empty({N, A2[n]}, bopt);
    auto h1n = torch::empty({N, H[n]}, bopt);
    auto h2n = torch::empty({N, H[n]}, bopt);
    op[n * 3 + 0] = outn.data_ptr();
    op[n * 3 + 1] = h1n.data_ptr();
    op[n * 3 + 2] = h2n.data_ptr();
    res.push_back(outn);
    res.push_back(h1n);
    res.push_back(h2n);
  }
  auto xb = xf32 ? torch::empty({N, O}, bopt) : x;
  auto stream = c10::hip::getCurrentHIPStream().stream();
  launch_mlp3_mfma_fwd2(x.data_ptr(), xf32 ? 1 : 0, wp, op, xb.data_ptr(), N,
                        (int)O, H, A2, (void*)stream);
  res.push_back(xb);
  return res;  // a_out, a_h1, a_h2, c_out, c_h1, c_h2, xb
}

std::vector<torch::Tensor> mlp3_mfma_bwd2(
    torch::Tensor a_dout, torch::Tensor a_h1, torch::Tensor a_h2,
    torch::Tensor a_w2, torch::Tensor a_w3, torch::Tensor c_dout,
    torch::Tensor c_h1, torch::Tensor c_h2, torch::Tensor c_w2,
    torch::Tensor c_w3) {
  const long N = a_dout.size(0);
  int H[2] = {(int)a_h1.size(1), (int)c_h1.size(1)};
  int A2[2] = {(int)a_dout.size(1), (int)c_dout.size(1)};
  TORCH_CHECK(a_dout.scalar_type() == torch::kBFloat16 &&
                  c_dout.scalar_type() == torch::kBFloat16,
              "bf16 douts");
  TORCH_CHECK(a_dout.is_contiguous() && c_dout.is_contiguous(), "contiguous");
  auto bopt = a_h1.options();
  const void* dp[10] = {a_dout.data_ptr(), a_h1.data_ptr(), a_h2.data_ptr(),
                        a_w2.data_ptr(),  a_w3.data_ptr(), c_dout.data_ptr(),
                        c_h1.data_ptr(),  c_h2.data_ptr(), c_w2.data_ptr(),
                        c_w3.data_ptr()};
  auto a_dh1 = torch::empty({N, H[0]}, bopt);
  auto a_dh2 = torch::empty({N, H[0]}, bopt);
  auto c_dh1 = torch::empty({N, H[1]}, bopt);
  auto c_dh2 = torch::empty({N, H[1]}, bopt);
  void* dh[4] = {a_dh1.data_ptr(), a_dh2.data_ptr(), c_dh1.data_ptr(),
                 c_dh2.data_ptr()};
  auto stream = c10::hip::getCurrentHIPStream().stream();
  launch_mlp3_mfma_bwd2(dp, dh, N, H, A2, (void*)stream);
  return {a_dh1, a_dh2, c_dh1, c_dh2};
}

// MFMA whole-MLP forward/backward (csrc/fused_mlp.hip v2): three
// matrix-core GEMMs + biases + tanh in one launch; dgrad chain + tanh'
// in one launch.  Replaces the eager hipBLASLt GEMM + tanh + cast
// chains of the PPO update phase.
bool mlp3_mfma_ok(long O, long H, long A2) {
  return H % 16 == 0 && H >= 16 && H <= 512 && O >= 1 && A2 >= 1 &&
         mlp3_mfma_lds_bytes((int)O, (int)H, (int)A2) <= 160 * 1024;
}

std::vector<torch::Tensor> mlp3_mfma_fwd(torch::Tensor x, torch::Tensor w1,
                                         torch::Tensor b1, torch::Tensor w2,
                                         torch::Tensor b2, torch::Tensor w3,
                                         torch::Tensor b3) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous(), "x cuda contiguous");
  const bool xf32 = x.scalar_type() == torch::kFloat32;
  TORCH_CHECK(xf32 || x.scalar_type() == torch::kBFloat16, "x fp32/bf16");
  TORCH_CHECK(w1.scalar_type() == torch::kBFloat16 &&
                  w2.scalar_type() == torch::kBFloat16 &&
                  w3.scalar_type() == torch::kBFloat16,
              "bf16 weights");
  const long N = x.size(0), O = x.size(1);
  const long H = w1.size(0), A2 = w3.size(0);
  TORCH_CHECK(w1.size(1) == O && w2.size(0) == H && w2.size(1) == H &&
                  w3.size(1) == H,
              "weight shape mismatch");
  TORCH_CHECK(mlp3_mfma_ok(O, H, A2), "mlp3_mfma: unsupported dims");
  auto bopt = w1.options();
  auto out = torch::empty({N, A2}, bopt);
  auto h1 = torch::empty({N, H}, bopt);
  auto h2 = torch::empty({N, H}, bopt);
  auto xb = xf32 ? torch::empty({N, O}, bopt) : x;
  auto stream = c10::hip::getCurrentHIPStream().stream();
  launch_mlp3_mfma_fwd(x.data_ptr(), xf32 ? 1 : 0, w1.data_ptr(),
                       b1.data_ptr(), w2.data_ptr(), b2.data_ptr(),
                       w3.data_ptr(), b3.data_ptr(), out.data_ptr(),
                       h1.data_ptr(), h2.data_ptr(), xb.data_ptr(), N, (int)O,
                       (int)H, (int)A2, (void*)stream);
  return {out, h1, h2, xb};
}

std::vector<torch::Tensor> mlp3_mfma_bwd(torch::Tensor dout, torch::Tensor h1,
                                         torch::Tensor h2, torch::Tensor w2,
                                         torch::Tensor w3) {
  TORCH_CHECK(dout.is_cuda() && dout.is_contiguous(), "dout cuda contiguous");
  const bool df32 = dout.scalar_type() == torch::kFloat32;
  const long N = dout.size(0), A2 = dout.size(1), H = h1.size(1);
  auto bopt = h1.options();
  auto dh1 = torch::empty({N, H}, bopt);
  auto dh2 = torch::empty({N, H}, bopt);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  launch_mlp3_mfma_bwd(dout.data_ptr(), df32 ? 1 : 0, h1.data_ptr(),
                       h2.data_ptr(), w2.data_ptr(), w3.data_ptr(),
                       dh1.data_ptr(), dh2.data_ptr(), N, (int)H, (int)A2,
                       (void*)stream);
  return {dh1, dh2};
}

// Same-network pair forward (GAE value/next_value in one launch; no
// gradients, scratch activations discarded).
std::vector<torch::Tensor> mlp3_mfma_fwdpair(torch::Tensor x0,
                                             torch::Tensor x1,
                                             std::vector<torch::Tensor> w) {
  TORCH_CHECK(x0.is_cuda() && x0.is_contiguous() && x1.is_contiguous(),
              "inputs cuda contiguous");
  TORCH_CHECK(x0.scalar_type() == x1.scalar_type() &&
                  x0.sizes() == x1.sizes(),
              "pair inputs must match");
  TORCH_CHECK(w.size() == 6, "6 weight tensors");
  const bool xf32 = x0.scalar_type() == torch::kFloat32;
  const long N = x0.size(0), O = x0.size(1);
  const long H = w[0].size(0), A2 = w[4].size(0);
  TORCH_CHECK(mlp3_mfma_ok(O, H, A2), "unsupported dims");
  const void* wp[6];
  for (int i = 0; i < 6; ++i) wp[i] = w[i].data_ptr();
  auto bopt = w[0].options();
  auto out0 = torch::empty({N, A2}, bopt);
  auto out1 = torch::empty({N, A2}, bopt);
  auto scratch = torch::empty({4 * N * H}, bopt);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  launch_mlp3_mfma_fwdpair(x0.data_ptr(), x1.data_ptr(), xf32 ? 1 : 0, wp,
                           out0.data_ptr(), out1.data_ptr(),
                           scratch.data_ptr(), N, (int)O, (int)H, (int)A2,
                           (void*)stream);
  return {out0, out1};
}

// Per-minibatch advantage mean/std in one launch pair (grid.y = slice)
torch::Tensor adv_stats_batch(torch::Tensor adv, long n_mb) {
  TORCH_CHECK(adv.is_cuda() && adv.is_contiguous() &&
                  adv.scalar_type() == torch::kFloat32,
              "adv fp32 cuda contiguous");
  const long n = adv.numel();
  TORCH_CHECK(n % n_mb == 0, "adv not divisible into minibatches");
  auto part = torch::empty({n_mb * 512}, adv.options());
  auto stats = torch::empty({n_mb, 2}, adv.options());
  auto stream = c10::hip::getCurrentHIPStream().stream();
  launch_adv_stats_batch(adv.data_ptr<float>(), part.data_ptr<float>(),
                         stats.data_ptr<float>(), n / n_mb, (int)n_mb,
                         (void*)stream);
  return stats;
}

// Whole-minibatch fused forward: actor+critic MLPs AND every PPO loss
// scalar in one launch + one finalize (csrc/fused_mlp.hip).
std::vector<torch::Tensor> acloss_fwd(
    torch::Tensor x, std::vector<torch::Tensor> aw,
    std::vector<torch::Tensor> cw, torch::Tensor action, torch::Tensor eps,
    torch::Tensor prev, torch::Tensor adv, torch::Tensor vtarget,
    torch::Tensor stats_in, double sp_bias, double lb, double lo, double hi,
    double ent_coeff, double crit_scale, bool normalize) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous(), "x cuda contiguous");
  const bool xf32 = x.scalar_type() == torch::kFloat32;
  const long N = x.size(0), O = x.size(1);
  int H[2] = {(int)aw[0].size(0), (int)cw[0].size(0)};
  int A2[2] = {(int)aw[4].size(0), (int)cw[4].size(0)};
  const int Aact = A2[0] / 2;
  TORCH_CHECK(A2[1] == 1 && A2[0] == 2 * Aact, "actor 2A, critic 1");
  TORCH_CHECK(mlp3_mfma_ok(O, H[0], A2[0]) && mlp3_mfma_ok(O, H[1], A2[1]),
              "unsupported dims");
  TORCH_CHECK(action.is_contiguous() && eps.is_contiguous() &&
                  prev.is_contiguous() && adv.is_contiguous() &&
                  vtarget.is_contiguous(),
              "loss inputs contiguous");
  auto bopt = aw[0].options();
  auto fopt = action.options();
  const void* wp[12];
  void* op[6];
  std::vector<torch::Tensor> res;
  for (int n = 0; n < 2; ++n) {
    auto& w = n == 0 ? aw : cw;
    for (int i = 0; i < 6; ++i) wp[n * 6 + i] = w[i].data_ptr();
    auto outn = torch::empty({N, A2[n]}, bopt);
    auto h1n = torch::empty({N, H[n]}, bopt);
    auto h2n = torch::empty({N, H[n]}, bopt);
    op[n * 3 + 0] = outn.data_ptr();
    op[n * 3 + 1] = h1n.data_ptr();
    op[n * 3 + 2] = h2n.data_ptr();
    res.push_back(outn);
    res.push_back(h1n);
    res.push_back(h2n);
  }
  auto xb = xf32 ? torch::empty({N, O}, bopt) : x;
  const int R = N >= 32768 ? 64 : 32;
  const int blocks = (int)((N + R - 1) / R);
  auto part = torch::empty({(long)blocks * 7 + 512}, fopt);
  float* outp[8];
  std::vector<torch::Tensor> scalars;
  for (int i = 0; i < 8; ++i) {
    scalars.push_back(torch::empty({}, fopt));
    outp[i] = scalars.back().data_ptr<float>();
  }
  const bool ext_stats = stats_in.numel() == 2;
  auto stats = ext_stats
                   ? stats_in
                   : (normalize ? torch::empty({2}, fopt)
                                : torch::empty({0}, fopt));
  auto stream = c10::hip::getCurrentHIPStream().stream();
  if (normalize && !ext_stats)
    launch_adv_stats(adv.data_ptr<float>(), part.data_ptr<float>(),
                     stats.data_ptr<float>(), N, (void*)stream);
  launch_mlp3_mfma_fwd2_loss(
      x.data_ptr(), xf32 ? 1 : 0, wp, op, xb.data_ptr(),
      action.data_ptr<float>(), eps.data_ptr<float>(),
      prev.data_ptr<float>(), adv.data_ptr<float>(),
      normalize ? stats.data_ptr<float>() : nullptr,
      vtarget.data_ptr<float>(), part.data_ptr<float>(), outp,
      (float)sp_bias, (float)lb, (float)lo, (float)hi, (float)ent_coeff,
      (float)crit_scale, N, (int)O, H, A2, Aact, (void*)stream);
  res.push_back(xb);
  for (auto& t : scalars) res.push_back(t);
  res.push_back(stats);
  // head, a_h1, a_h2, value, c_h1, c_h2, xb, 8 scalars, stats
  return res;
}

std::vector<torch::Tensor> acloss_bwd(
    torch::Tensor head, torch::Tensor value, torch::Tensor a_h1,
    torch::Tensor a_h2, torch::Tensor a_w2, torch::Tensor a_w3,
    torch::Tensor c_h1, torch::Tensor c_h2, torch::Tensor c_w2,
    torch::Tensor c_w3, torch::Tensor action, torch::Tensor eps,
    torch::Tensor prev, torch::Tensor adv, torch::Tensor stats,
    torch::Tensor vtarget, torch::Tensor gobj, torch::Tensor gent,
    torch::Tensor gact, torch::Tensor gcrit, torch::Tensor gtot,
    double sp_bias, double lb, double lo, double hi, double ent_coeff,
    double crit_scale) {
  const long N = head.size(0);
  int H[2] = {(int)a_h1.size(1), (int)c_h1.size(1)};
  int A2[2] = {(int)head.size(1), 1};
  const int Aact = A2[0] / 2;
  auto bopt = head.options();
  auto dhead = torch::empty_like(head);
  auto dvalue = torch::empty_like(value);
  const void* hw[8] = {a_h1.data_ptr(), a_h2.data_ptr(), a_w2.data_ptr(),
                       a_w3.data_ptr(), c_h1.data_ptr(), c_h2.data_ptr(),
                       c_w2.data_ptr(), c_w3.data_ptr()};
  auto a_dh1 = torch::empty({N, H[0]}, bopt);
  auto a_dh2 = torch::empty({N, H[0]}, bopt);
  auto c_dh1 = torch::empty({N, H[1]}, bopt);
  auto c_dh2 = torch::empty({N, H[1]}, bopt);
  void* dh[4] = {a_dh1.data_ptr(), a_dh2.data_ptr(), c_dh1.data_ptr(),
                 c_dh2.data_ptr()};
  auto stream = c10::hip::getCurrentHIPStream().stream();
  launch_mlp3_mfma_bwd2_loss(
      hw, dh, action.data_ptr<float>(), eps.data_ptr<float>(),
      prev.data_ptr<float>(), adv.data_ptr<float>(),
      stats.numel() ? stats.data_ptr<float>() : nullptr,
      vtarget.data_ptr<float>(),
      gobj.numel() ? gobj.data_ptr<float>() : nullptr,
      gent.numel() ? gent.data_ptr<float>() : nullptr,
      gact.numel() ? gact.data_ptr<float>() : nullptr,
      gcrit.numel() ? gcrit.data_ptr<float>() : nullptr,
      gtot.numel() ? gtot.data_ptr<float>() : nullptr, dhead.data_ptr(),
      dvalue.data_ptr(), head.data_ptr(), value.data_ptr(), (float)sp_bias,
      (float)lb, (float)lo, (float)hi, (float)ent_coeff, (float)crit_scale,
      N, H, A2, Aact, (void*)stream);
  return {dhead, dvalue, a_dh1, a_dh2, c_dh1, c_dh2};
}

// Mega-fused TanhNormal head loss: raw actor-head output [N, 2A] ->
// (out[5] = {loss_objective, ESS/N, clip_fraction, entropy_mean,
// loss_entropy}, stats).  See csrc/loss_ops.hip.
std::vector<torch::Tensor> ppo_head_fwd(torch::Tensor head,
                                        torch::Tensor action,
                                        torch::Tensor eps, torch::Tensor prev,
                                        torch::Tensor adv,
                                        torch::Tensor value,
                                        torch::Tensor vtarget,
                                        double sp_bias, double lb, double lo,
                                        double hi, double ent_coeff,
                                        double crit_scale, bool normalize) {
  TORCH_CHECK(head.is_cuda() && head.is_contiguous(), "head cuda contiguous");
  const bool bf16 = head.scalar_type() == torch::kBFloat16;
  TORCH_CHECK(bf16 || head.scalar_type() == torch::kFloat32,
              "head fp32/bf16");
  TORCH_CHECK(action.scalar_type() == torch::kFloat32 &&
                  eps.scalar_type() == torch::kFloat32 &&
                  prev.scalar_type() == torch::kFloat32 &&
                  adv.scalar_type() == torch::kFloat32,
              "fp32 data inputs");
  TORCH_CHECK(action.is_contiguous() && eps.is_contiguous() &&
                  prev.is_contiguous() && adv.is_contiguous(),
              "contiguous");
  const long N = head.size(0);
  const int A = (int)(head.size(1) / 2);
  TORCH_CHECK(head.size(1) == 2 * A && action.numel() == N * A &&
                  eps.numel() == N * A && prev.numel() == N &&
                  adv.numel() == N,
              "shape mismatch");
  const bool has_crit = value.numel() > 0;
  if (has_crit) {
    TORCH_CHECK(value.numel() == N && vtarget.numel() == N &&
                    value.scalar_type() == head.scalar_type() &&
                    vtarget.scalar_type() == torch::kFloat32 &&
                    value.is_contiguous() && vtarget.is_contiguous(),
                "ppo_head_fwd: critic inputs");
  }
  auto opt = action.options();
  auto part = torch::empty({256 * 6}, opt);
  std::vector<torch::Tensor> outs;
  float* outp[8];
  for (int i = 0; i < 8; ++i) {
    outs.push_back(torch::empty({}, opt));
    outp[i] = outs.back().data_ptr<float>();
  }
  auto stats = normalize ? torch::empty({2}, opt) : torch::empty({0}, opt);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  if (normalize)
    launch_adv_stats(adv.data_ptr<float>(), part.data_ptr<float>(),
                     stats.data_ptr<float>(), N, (void*)stream);
  launch_ppo_head_fwd(head.data_ptr(), action.data_ptr<float>(),
                      eps.data_ptr<float>(), prev.data_ptr<float>(),
                      adv.data_ptr<float>(),
                      normalize ? stats.data_ptr<float>() : nullptr,
                      has_crit ? value.data_ptr() : nullptr,
                      has_crit ? vtarget.data_ptr<float>() : nullptr,
                      part.data_ptr<float>(), outp, (float)sp_bias,
                      (float)lb, (float)lo, (float)hi, (float)ent_coeff,
                      (float)crit_scale, N, A, bf16 ? 1 : 0, (void*)stream);
  outs.push_back(stats);
  return outs;  // 8 scalars + stats
}

std::vector<torch::Tensor> ppo_head_bwd(
    torch::Tensor head, torch::Tensor action, torch::Tensor eps,
    torch::Tensor prev, torch::Tensor adv, torch::Tensor stats,
    torch::Tensor value, torch::Tensor vtarget, torch::Tensor gobj,
    torch::Tensor gent, torch::Tensor gact, torch::Tensor gcrit,
    torch::Tensor gtot, double sp_bias, double lb, double lo, double hi,
    double ent_coeff, double crit_scale) {
  const bool bf16 = head.scalar_type() == torch::kBFloat16;
  const long N = head.size(0);
  const int A = (int)(head.size(1) / 2);
  const bool has_crit = value.numel() > 0;
  auto dhead = torch::empty_like(head);
  auto dvalue = has_crit ? torch::empty_like(value)
                         : torch::empty({0}, head.options());
  auto stream = c10::hip::getCurrentHIPStream().stream();
  launch_ppo_head_bwd(
      head.data_ptr(), action.data_ptr<float>(), eps.data_ptr<float>(),
      prev.data_ptr<float>(), adv.data_ptr<float>(),
      stats.numel() ? stats.data_ptr<float>() : nullptr,
      has_crit ? value.data_ptr() : nullptr,
      has_crit ? vtarget.data_ptr<float>() : nullptr,
      gobj.numel() ? gobj.data_ptr<float>() : nullptr,
      gent.numel() ? gent.data_ptr<float>() : nullptr,
      gact.numel() ? gact.data_ptr<float>() : nullptr,
      gcrit.numel() ? gcrit.data_ptr<float>() : nullptr,
      gtot.numel() ? gtot.data_ptr<float>() : nullptr, dhead.data_ptr(),
      has_crit ? dvalue.data_ptr() : nullptr, (float)sp_bias, (float)lb,
      (float)lo, (float)hi, (float)ent_coeff, (float)crit_scale, N, A,
      bf16 ? 1 : 0, (void*)stream);
  return {dhead, dvalue};
}
#endif  // RL_AMD_WITH_HIP

// ---------------------------------------------------------------------------
// safetanh / safeatanh (reference torchrl/csrc/utils.cpp:9-48)
// ---------------------------------------------------------------------------
torch::Tensor safetanh_fwd(torch::Tensor x, double eps) {
  auto out = x.tanh();
  return out.clamp(-1.0 + eps, 1.0 - eps);
}

torch::Tensor safeatanh_fwd(torch::Tensor y, double eps) {
  return y.clamp(-1.0 + eps, 1.0 - eps).atanh();
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "rl_amd native ops (CPU segment trees + CDNA4 HIP kernels)";
  m.attr("COMPILED_WITH_HIP") =
#ifdef RL_AMD_WITH_HIP
      true;
#else
      false;
#endif

  rl_amd::define_segment_trees(m);
  m.def("safetanh", &safetanh_fwd, "clamped tanh");
  m.def("safeatanh", &safeatanh_fwd, "clamped atanh");

#ifdef RL_AMD_WITH_HIP
  m.def("gae", &gae, "fused GAE scan (HIP)");
  m.def("synthetic_env_step", &synthetic_env_step,
        "fused synthetic-MuJoCo env transition (HIP)");
  m.def("fused_rollout", &fused_rollout,
        "entire T-step rollout in one launch (HIP)");
  m.def("mlp3_fwd", &mlp3_fwd, "fused 3-layer MLP forward (HIP)");
  m.def("mlp3_bwd", &mlp3_bwd, "fused MLP backward dgrad chain (HIP)");
  m.def("tanh_normal_entropy", &tanh_normal_entropy,
        "fused reparameterized TanhNormal MC entropy (HIP)");
  m.def("tanh_normal_entropy_bwd", &tanh_normal_entropy_bwd,
        "fused entropy backward (HIP)");
  m.def("tanh_normal_logprob", &tanh_normal_logprob,
        "fused TanhNormal log-prob forward (HIP)");
  m.def("tanh_normal_logprob_bwd", &tanh_normal_logprob_bwd,
        "fused TanhNormal log-prob backward (HIP)");
  m.def("fused_actor_into", &fused_actor_into,
        "store-direct fused actor (HIP)");
  m.def("ppo_clip_fwd", &ppo_clip_fwd,
        "fused ClipPPO objective + diagnostics forward (HIP)");
  m.def("ppo_clip_bwd", &ppo_clip_bwd, "fused ClipPPO backward (HIP)");
  m.def("smooth_l1_fwd", &smooth_l1_fwd,
        "fused mean smooth-L1 critic loss forward (HIP)");
  m.def("smooth_l1_bwd", &smooth_l1_bwd,
        "fused smooth-L1 backward (HIP)");
  m.def("mlp3_mfma_ok", &mlp3_mfma_ok, "MFMA MLP3 shape eligibility");
  m.def("wgrad_splitk3", &wgrad_splitk3,
        "batched 3-layer split-K wgrad (HIP)");
  m.def("wgrad_splitk_batch", &wgrad_splitk_batch,
        "batched 1..6-layer split-K wgrad (HIP)");
  m.def("wgrad_splitk_batch_sq", &wgrad_splitk_batch_sq,
        "batched wgrad emitting grad-sumsq partials (HIP)");
  m.def("wgrad_clip_finalize", &wgrad_clip_finalize,
        "clip coefficient from wgrad sq partials (HIP)");
  m.def("mlp3_mfma_fwd2", &mlp3_mfma_fwd2,
        "dual-network MFMA MLP forward (HIP)");
  m.def("mlp3_mfma_bwd2", &mlp3_mfma_bwd2,
        "dual-network MFMA dgrad chain (HIP)");
  m.def("mlp3_mfma_fwdpair", &mlp3_mfma_fwdpair,
        "same-network pair forward (HIP)");
  m.def("adv_stats_batch", &adv_stats_batch,
        "per-minibatch advantage mean/std, one launch pair (HIP)");
  m.def("acloss_fwd", &acloss_fwd,
        "actor+critic MLPs + every PPO loss scalar in one launch (HIP)");
  m.def("acloss_bwd", &acloss_bwd,
        "merged d(head)/d(value) + dgrad chains in one launch (HIP)");
  m.def("fused_grad_clip_coef", &fused_grad_clip_coef,
        "single-kernel global grad-norm clip coefficient (HIP)");
  m.def("multi_gather", &multi_gather,
        "batched row gather over up to 8 tensors (HIP)");
  m.def("multi_shuffle", &multi_shuffle,
        "batched Feistel-permutation shuffle (HIP)");
  m.def("mlp3_mfma_fwd", &mlp3_mfma_fwd,
        "MFMA whole-MLP forward: 3 GEMMs + tanh in one launch (HIP)");
  m.def("mlp3_mfma_bwd", &mlp3_mfma_bwd,
        "MFMA dgrad chain + tanh' in one launch (HIP)");
  m.def("ppo_head_fwd", &ppo_head_fwd,
        "mega-fused TanhNormal head -> PPO losses forward (HIP)");
  m.def("ppo_head_bwd", &ppo_head_bwd,
        "mega-fused head-loss backward (HIP)");
  m.def("synthetic_env_step_into", &synthetic_env_step_into,
        "store-direct fused env transition with auto-reset (HIP)");
  m.def("wgrad_splitk", &wgrad_splitk,
        "split-K skinny weight gradient dW=dY^T X (HIP)");
  m.def("revscan", &revscan, "generic reverse linear-recurrence scan (HIP)");
  m.def("vtrace", &vtrace, "fused V-trace (HIP)");
  m.def("tree_scan_lower_bound", &tree_scan_lower_bound,
        "sum-tree inverse-CDF descent (HIP)");
  m.def("tree_update", &tree_update, "segment-tree leaf update+recompute (HIP)");
  m.def("gru_fused", &gru_fused, "fused GRU forward scan with resets (HIP)");
  m.def("lstm_fused", &lstm_fused, "fused LSTM forward scan with resets (HIP)");
  m.def("gru_train_fwd", &gru_train_fwd,
        "differentiable GRU forward scan, any H (HIP)");
  m.def("gru_bwd", &gru_bwd,
        "GRU reverse-time backward scan with gate recompute (HIP)");
  m.def("lstm_train_fwd", &lstm_train_fwd,
        "LSTM forward scan emitting per-step cell states (HIP)");
  m.def("lstm_bwd", &lstm_bwd,
        "LSTM reverse-time backward scan with gate recompute (HIP)");
  m.def("fused_actor", &fused_actor,
        "fused MLP+TanhNormal actor forward (HIP)");
#endif
}
