// Python bindings for the stoix_amd gfx950 kernels.
//
// Pure-C++ translation layer: validates torch tensors, extracts raw
// pointers + the current HIP stream, and calls the extern "C" launchers
// defined in the .hip files. Every entry point takes the stream from
// PyTorch's current stream so the kernels compose with torch ops and with
// hip graph capture.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include <cstdint>

extern "C" {
void launch_cartpole_step(float*, const long*, int*, float*, int*, float*,
                          int*, float*, float*, float*, float*, unsigned char*,
                          unsigned char*, int, int, uint64_t, unsigned int*,
                          unsigned int, int, void*);
void launch_ant_step(float*, const float*, int*, float*, int*, float*, int*,
                     float*, float*, float*, float*, unsigned char*,
                     unsigned char*, int, int, uint64_t, unsigned int*,
                     unsigned int, int, void*);
void launch_humanoid_step(float*, const float*, int*, float*, int*, float*,
                          int*, float*, float*, float*, float*,
                          unsigned char*, unsigned char*, int, int, uint64_t,
                          unsigned int*, unsigned int, int, void*);
void launch_ant_reset(float*, int, uint64_t, uint32_t, void*);
void launch_gae(const float*, const float*, const float*, const float*,
                const unsigned char*, float*, float*, int, int, float, void*);
void launch_lambda_returns(const float*, const float*, const float*, float*,
                           int, int, float, void*);
void launch_vtrace(const float*, const float*, const float*, const float*,
                   const float*, float*, float*, int, int, float, float,
                   float, void*);
void launch_offpolicy_returns(const float*, const float*, const float*,
                              const float*, const float*, float*, int, int,
                              void*);
void launch_fused_adam(float*, const float*, float*, float*, float*, long*,
                       long, float, float, float, float, float, void*);
void launch_fused_adam_bf16(float*, const void*, float*, float*, float*,
                            long*, void*, long, float, float, float, float,
                            float, float, int, void*);
void launch_polyak(const float*, float*, long, float, void*);
void launch_mfma_probe(const void*, const void*, float*, float*, void*);
void launch_policy_value_step(const float*, const void*, const float*,
                              const void*, const float*, const void*,
                              const float*, const void*, const float*,
                              const void*, const float*, const void*,
                              const float*, float*, float*, float*, float*,
                              const float*, const float*, int, int, int, int,
                              float, float, float, float, int, uint64_t,
                              unsigned int*, unsigned int, int, void*);
void launch_value_forward(const float*, const void*, const float*,
                          const void*, const float*, const void*,
                          const float*, float*, const float*, const float*,
                          int, int, int, void*);
void launch_rollout_step_ant(float*, float*, int*, float*, int*, float*,
                             int*, const void*, const float*, const void*,
                             const float*, const void*, const float*,
                             const void*, const float*, const void*,
                             const float*, const void*, const float*, float*,
                             float*, float*, float*, float*, float*, float*,
                             unsigned char*, int, int, int, int, int, float,
                             float, float, float, uint64_t, uint64_t,
                             unsigned int*, unsigned int*, unsigned int,
                             void*);
void launch_linear_silu(const void*, const void*, const float*, void*,
                        void*, int, int, int, int, void*);
void launch_silu_fwd(const void*, void*, long, void*);
void launch_silu_bwd(const void*, const void*, void*, long, void*);
void launch_ppo_gather(const long*, int, const float*, int, int,
                       const float*, int,
                       const float*, const float*, const float*, const float*,
                       void*, float*, float*, float*, float*, float*,
                       const float*, const float*, void*);
void launch_ppo_head_loss(const void*, const void*, const float*,
                          const float*, const float*, const float*,
                          const float*, void*, void*, void*, float*, int,
                          int, float, float, float, float, float, float,
                          float, uint64_t, unsigned int*, unsigned int, int,
                          void*);
void launch_wgrad(const void*, const void*, float*, long, long, long, int,
                  int, int, int, int, int, void*);
void launch_slab_reduce(float*, void*, long, long, float*, long*, void*);
void launch_bump_add(unsigned int*, unsigned int, void*);
void launch_tr16_probe(const void*, float*, int, void*);
void launch_sumtree_update(float*, const long*, const float*, long, int, int,
                           void*);
void launch_sumtree_sample(const float*, const float*, long*, long, int, int,
                           int, void*);
void launch_policy_value_step_disc(const float*, const void*, const float*,
                                   const void*, const float*, const void*,
                                   const float*, const void*, const float*,
                                   const void*, const float*, const void*,
                                   const float*, float*, long*, float*,
                                   float*, const float*, const float*, int,
                                   int, int, int, int, uint64_t,
                                   unsigned int*, unsigned int, int, void*);
void launch_ppo_head_loss_disc(const void*, const void*, const long*,
                               const float*, const float*, const float*,
                               const float*, void*, void*, void*, float*,
                               int, int, float, float, float, void*);
void launch_ppo_gather_disc(const long*, int, const float*, int, int,
                            const long*, const float*, const float*,
                            const float*, const float*, void*, long*, float*,
                            float*, float*, float*, const float*,
                            const float*, void*);
void launch_rnn_scan(const float*, const void*, const float*,
                     const unsigned char*, const float*, const float*,
                     float*, float*, float*, int, int, int, int, void*);
void launch_snake_step(int*, long*, long*, long*, long*, int*, const long*,
                       int*, float*, int*, float*, int*, float*, float*,
                       float*, float*, unsigned char*, unsigned char*, int,
                       int, uint64_t, unsigned int*, unsigned int, int,
                       void*);
}

namespace {

void* cur_stream() {
  return (void*)at::hip::getCurrentHIPStream().stream();
}

#define CHK(t, ty)                                                     \
  TORCH_CHECK((t).is_cuda(), #t " must be on GPU");                    \
  TORCH_CHECK((t).is_contiguous(), #t " must be contiguous");          \
  TORCH_CHECK((t).scalar_type() == ty, #t " has wrong dtype");

void cartpole_step(torch::Tensor state, torch::Tensor action,
                   torch::Tensor step_count, torch::Tensor ep_return,
                   torch::Tensor ep_length, torch::Tensor last_ep_return,
                   torch::Tensor last_ep_length, torch::Tensor obs_out,
                   torch::Tensor next_obs_out, torch::Tensor reward_out,
                   torch::Tensor discount_out, torch::Tensor steptype_out,
                   torch::Tensor done_out, int64_t max_episode_steps,
                   int64_t seed, torch::Tensor draw_buf, int64_t draw_offset,
                   int64_t do_bump) {
  CHK(state, torch::kFloat32);
  CHK(action, torch::kInt64);
  int B = state.size(0);
  launch_cartpole_step(
      state.data_ptr<float>(), action.data_ptr<long>(),
      step_count.data_ptr<int>(), ep_return.data_ptr<float>(),
      ep_length.data_ptr<int>(), last_ep_return.data_ptr<float>(),
      last_ep_length.data_ptr<int>(), obs_out.data_ptr<float>(),
      next_obs_out.data_ptr<float>(), reward_out.data_ptr<float>(),
      discount_out.data_ptr<float>(), steptype_out.data_ptr<unsigned char>(),
      done_out.data_ptr<unsigned char>(), B, (int)max_episode_steps,
      (uint64_t)seed, (unsigned int*)draw_buf.data_ptr<int>(),
      (unsigned int)draw_offset, (int)do_bump, cur_stream());
}

void ant_step(torch::Tensor state, torch::Tensor action,
              torch::Tensor step_count, torch::Tensor ep_return,
              torch::Tensor ep_length, torch::Tensor last_ep_return,
              torch::Tensor last_ep_length, torch::Tensor obs_out,
              torch::Tensor next_obs_out, torch::Tensor reward_out,
              torch::Tensor discount_out, torch::Tensor steptype_out,
              torch::Tensor done_out, int64_t max_episode_steps, int64_t seed,
              torch::Tensor draw_buf, int64_t draw_offset, int64_t do_bump) {
  CHK(state, torch::kFloat32);
  CHK(action, torch::kFloat32);
  int B = state.size(0);
  launch_ant_step(
      state.data_ptr<float>(), action.data_ptr<float>(),
      step_count.data_ptr<int>(), ep_return.data_ptr<float>(),
      ep_length.data_ptr<int>(), last_ep_return.data_ptr<float>(),
      last_ep_length.data_ptr<int>(), obs_out.data_ptr<float>(),
      next_obs_out.data_ptr<float>(), reward_out.data_ptr<float>(),
      discount_out.data_ptr<float>(), steptype_out.data_ptr<unsigned char>(),
      done_out.data_ptr<unsigned char>(), B, (int)max_episode_steps,
      (uint64_t)seed, (unsigned int*)draw_buf.data_ptr<int>(),
      (unsigned int)draw_offset, (int)do_bump, cur_stream());
}

void humanoid_step(torch::Tensor state, torch::Tensor action,
                   torch::Tensor step_count, torch::Tensor ep_return,
                   torch::Tensor ep_length, torch::Tensor last_ep_return,
                   torch::Tensor last_ep_length, torch::Tensor obs_out,
                   torch::Tensor next_obs_out, torch::Tensor reward_out,
                   torch::Tensor discount_out, torch::Tensor steptype_out,
                   torch::Tensor done_out, int64_t max_episode_steps,
                   int64_t seed, torch::Tensor draw_buf, int64_t draw_offset,
                   int64_t do_bump) {
  CHK(state, torch::kFloat32);
  CHK(action, torch::kFloat32);
  int B = state.size(0);
  launch_humanoid_step(
      state.data_ptr<float>(), action.data_ptr<float>(),
      step_count.data_ptr<int>(), ep_return.data_ptr<float>(),
      ep_length.data_ptr<int>(), last_ep_return.data_ptr<float>(),
      last_ep_length.data_ptr<int>(), obs_out.data_ptr<float>(),
      next_obs_out.data_ptr<float>(), reward_out.data_ptr<float>(),
      discount_out.data_ptr<float>(), steptype_out.data_ptr<unsigned char>(),
      done_out.data_ptr<unsigned char>(), B, (int)max_episode_steps,
      (uint64_t)seed, (unsigned int*)draw_buf.data_ptr<int>(),
      (unsigned int)draw_offset, (int)do_bump, cur_stream());
}

void ant_reset(torch::Tensor state, int64_t seed, int64_t draw) {
  CHK(state, torch::kFloat32);
  launch_ant_reset(state.data_ptr<float>(), state.size(0), (uint64_t)seed,
                   (uint32_t)draw, cur_stream());
}

void gae(torch::Tensor r_t, torch::Tensor discount_t, torch::Tensor v_tm1,
         torch::Tensor v_t, torch::Tensor trunc_t, torch::Tensor adv_out,
         torch::Tensor target_out, double lambda_) {
  CHK(r_t, torch::kFloat32);
  int T = r_t.size(0), B = r_t.size(1);
  const unsigned char* tr =
      trunc_t.numel() > 0 ? trunc_t.data_ptr<unsigned char>() : nullptr;
  launch_gae(r_t.data_ptr<float>(), discount_t.data_ptr<float>(),
             v_tm1.data_ptr<float>(), v_t.data_ptr<float>(), tr,
             adv_out.data_ptr<float>(), target_out.data_ptr<float>(), T, B,
             (float)lambda_, cur_stream());
}

void lambda_returns(torch::Tensor r_t, torch::Tensor discount_t,
                    torch::Tensor v_t, torch::Tensor out, double lambda_) {
  CHK(r_t, torch::kFloat32);
  launch_lambda_returns(r_t.data_ptr<float>(), discount_t.data_ptr<float>(),
                        v_t.data_ptr<float>(), out.data_ptr<float>(),
                        r_t.size(0), r_t.size(1), (float)lambda_,
                        cur_stream());
}

void vtrace(torch::Tensor v_tm1, torch::Tensor v_t, torch::Tensor r_t,
            torch::Tensor discount_t, torch::Tensor rho_tm1,
            torch::Tensor errors_out, torch::Tensor pg_adv_out, double lambda_,
            double clip_rho, double clip_pg_rho) {
  CHK(r_t, torch::kFloat32);
  launch_vtrace(v_tm1.data_ptr<float>(), v_t.data_ptr<float>(),
                r_t.data_ptr<float>(), discount_t.data_ptr<float>(),
                rho_tm1.data_ptr<float>(), errors_out.data_ptr<float>(),
                pg_adv_out.data_ptr<float>(), r_t.size(0), r_t.size(1),
                (float)lambda_, (float)clip_rho, (float)clip_pg_rho,
                cur_stream());
}

void offpolicy_returns(torch::Tensor q_t, torch::Tensor v_t, torch::Tensor r_t,
                       torch::Tensor discount_t, torch::Tensor c_t,
                       torch::Tensor out) {
  CHK(r_t, torch::kFloat32);
  launch_offpolicy_returns(q_t.data_ptr<float>(), v_t.data_ptr<float>(),
                           r_t.data_ptr<float>(), discount_t.data_ptr<float>(),
                           c_t.data_ptr<float>(), out.data_ptr<float>(),
                           r_t.size(0), r_t.size(1), cur_stream());
}

void fused_adam(torch::Tensor param, torch::Tensor grad, torch::Tensor exp_avg,
                torch::Tensor exp_avg_sq, torch::Tensor sqnorm,
                torch::Tensor step_t, double lr, double beta1, double beta2,
                double eps, double max_norm) {
  CHK(param, torch::kFloat32);
  CHK(grad, torch::kFloat32);
  launch_fused_adam(param.data_ptr<float>(), grad.data_ptr<float>(),
                    exp_avg.data_ptr<float>(), exp_avg_sq.data_ptr<float>(),
                    sqnorm.data_ptr<float>(), step_t.data_ptr<long>(),
                    param.numel(), (float)lr, (float)beta1, (float)beta2,
                    (float)eps, (float)max_norm, cur_stream());
}

void polyak(torch::Tensor online, torch::Tensor target, double tau) {
  CHK(online, torch::kFloat32);
  launch_polyak(online.data_ptr<float>(), target.data_ptr<float>(),
                online.numel(), (float)tau, cur_stream());
}

void fused_adam_bf16(torch::Tensor param, torch::Tensor grad,
                     torch::Tensor exp_avg, torch::Tensor exp_avg_sq,
                     torch::Tensor sqnorm, torch::Tensor step_t,
                     torch::Tensor param_bf16, double lr, double beta1,
                     double beta2, double eps, double max_norm,
                     double grad_scale, int64_t do_prologue) {
  CHK(param, torch::kFloat32);
  CHK(grad, torch::kBFloat16);
  void* pbf = param_bf16.numel() > 0 ? param_bf16.data_ptr() : nullptr;
  launch_fused_adam_bf16(param.data_ptr<float>(), grad.data_ptr(),
                         exp_avg.data_ptr<float>(),
                         exp_avg_sq.data_ptr<float>(),
                         sqnorm.data_ptr<float>(), step_t.data_ptr<long>(),
                         pbf, param.numel(), (float)lr, (float)beta1,
                         (float)beta2, (float)eps, (float)max_norm,
                         (float)grad_scale, (int)do_prologue, cur_stream());
}

void mfma_probe(torch::Tensor A, torch::Tensor B, torch::Tensor D0,
                torch::Tensor D1) {
  CHK(A, torch::kBFloat16);
  launch_mfma_probe(A.data_ptr(), B.data_ptr(), D0.data_ptr<float>(),
                    D1.data_ptr<float>(), cur_stream());
}

const float* fptr_or_null(const torch::Tensor& t) {
  return t.numel() > 0 ? t.data_ptr<float>() : nullptr;
}

void policy_value_step(torch::Tensor obs, torch::Tensor W1a, torch::Tensor b1a,
                       torch::Tensor W2a, torch::Tensor b2a, torch::Tensor Wha,
                       torch::Tensor bha, torch::Tensor W1c, torch::Tensor b1c,
                       torch::Tensor W2c, torch::Tensor b2c, torch::Tensor Wvc,
                       torch::Tensor bvc, torch::Tensor obs_mirror,
                       torch::Tensor action_out, torch::Tensor logp_out,
                       torch::Tensor value_out, torch::Tensor nmean,
                       torch::Tensor nvar, double min_scale, double aff_scale,
                       double aff_shift, double log_aff_scale, int64_t greedy,
                       int64_t seed, torch::Tensor draw_buf,
                       int64_t draw_offset, int64_t do_bump) {
  CHK(obs, torch::kFloat32);
  CHK(W1a, torch::kBFloat16);
  CHK(W2a, torch::kBFloat16);
  int B = obs.size(0), OBS = obs.size(1);
  int HID = W2a.size(0);
  int ACT = action_out.size(1);
  TORCH_CHECK(HID == 512 || HID == 256 || HID == 128,
              "fused MLP supports HID 128/256/512");
  TORCH_CHECK(ACT <= 8, "fused policy head supports ACT <= 8");
  TORCH_CHECK(OBS <= 128, "fused MLP supports OBS <= 128");
  float* om = obs_mirror.numel() > 0 ? obs_mirror.data_ptr<float>() : nullptr;
  unsigned int* db = draw_buf.numel() > 0
                         ? (unsigned int*)draw_buf.data_ptr<int>()
                         : nullptr;
  launch_policy_value_step(
      obs.data_ptr<float>(), W1a.data_ptr(), b1a.data_ptr<float>(),
      W2a.data_ptr(), b2a.data_ptr<float>(), Wha.data_ptr(),
      bha.data_ptr<float>(), W1c.data_ptr(), b1c.data_ptr<float>(),
      W2c.data_ptr(), b2c.data_ptr<float>(), Wvc.data_ptr(),
      bvc.data_ptr<float>(), om, action_out.data_ptr<float>(),
      logp_out.data_ptr<float>(), value_out.data_ptr<float>(),
      fptr_or_null(nmean), fptr_or_null(nvar), B, OBS, ACT, HID,
      (float)min_scale, (float)aff_scale, (float)aff_shift,
      (float)log_aff_scale, (int)greedy, (uint64_t)seed, db,
      (unsigned int)draw_offset, (int)do_bump, cur_stream());
}

void policy_value_step_disc(
    torch::Tensor obs, torch::Tensor W1a, torch::Tensor b1a,
    torch::Tensor W2a, torch::Tensor b2a, torch::Tensor Wha,
    torch::Tensor bha, torch::Tensor W1c, torch::Tensor b1c,
    torch::Tensor W2c, torch::Tensor b2c, torch::Tensor Wvc,
    torch::Tensor bvc, torch::Tensor obs_mirror, torch::Tensor action_out,
    torch::Tensor logp_out, torch::Tensor value_out, torch::Tensor nmean,
    torch::Tensor nvar, int64_t num_actions, int64_t greedy, int64_t seed,
    torch::Tensor draw_buf, int64_t draw_offset, int64_t do_bump) {
  CHK(obs, torch::kFloat32);
  CHK(W1a, torch::kBFloat16);
  CHK(action_out, torch::kInt64);
  int B = obs.size(0), OBS = obs.size(1);
  int HID = W2a.size(0);
  int ACT = (int)num_actions;
  TORCH_CHECK(HID == 512 || HID == 256 || HID == 128,
              "fused MLP supports HID 128/256/512");
  TORCH_CHECK(ACT <= 16, "fused categorical head supports ACT <= 16");
  TORCH_CHECK(OBS <= 128, "fused MLP supports OBS <= 128");
  float* om = obs_mirror.numel() > 0 ? obs_mirror.data_ptr<float>() : nullptr;
  unsigned int* db = draw_buf.numel() > 0
                         ? (unsigned int*)draw_buf.data_ptr<int>()
                         : nullptr;
  launch_policy_value_step_disc(
      obs.data_ptr<float>(), W1a.data_ptr(), b1a.data_ptr<float>(),
      W2a.data_ptr(), b2a.data_ptr<float>(), Wha.data_ptr(),
      bha.data_ptr<float>(), W1c.data_ptr(), b1c.data_ptr<float>(),
      W2c.data_ptr(), b2c.data_ptr<float>(), Wvc.data_ptr(),
      bvc.data_ptr<float>(), om, action_out.data_ptr<long>(),
      logp_out.data_ptr<float>(), value_out.data_ptr<float>(),
      fptr_or_null(nmean), fptr_or_null(nvar), B, OBS, ACT, HID, (int)greedy,
      (uint64_t)seed, db, (unsigned int)draw_offset, (int)do_bump,
      cur_stream());
}

void ppo_head_loss_disc(torch::Tensor heads, torch::Tensor v_in,
                        torch::Tensor action, torch::Tensor old_logp,
                        torch::Tensor old_value, torch::Tensor adv,
                        torch::Tensor targets, torch::Tensor dhead,
                        torch::Tensor dv, torch::Tensor dv16,
                        torch::Tensor metrics, int64_t num_actions,
                        double clip_eps, double ent_coef, double vf_coef) {
  CHK(heads, torch::kBFloat16);
  CHK(action, torch::kInt64);
  int B = heads.size(0);
  launch_ppo_head_loss_disc(
      heads.data_ptr(), v_in.data_ptr(), action.data_ptr<long>(),
      old_logp.data_ptr<float>(), old_value.data_ptr<float>(),
      adv.data_ptr<float>(), targets.data_ptr<float>(), dhead.data_ptr(),
      dv.data_ptr(), dv16.numel() > 0 ? dv16.data_ptr() : nullptr,
      metrics.numel() > 0 ? metrics.data_ptr<float>() : nullptr, B,
      (int)num_actions, (float)clip_eps, (float)ent_coef, (float)vf_coef,
      cur_stream());
}

void ppo_gather_disc(torch::Tensor idx, torch::Tensor obs,
                     torch::Tensor action, torch::Tensor logp,
                     torch::Tensor value, torch::Tensor adv,
                     torch::Tensor targets, torch::Tensor obs_out,
                     torch::Tensor action_out, torch::Tensor logp_out,
                     torch::Tensor value_out, torch::Tensor adv_out,
                     torch::Tensor targets_out, torch::Tensor nmean,
                     torch::Tensor nvar) {
  CHK(idx, torch::kInt64);
  CHK(obs, torch::kFloat32);
  CHK(obs_out, torch::kBFloat16);
  CHK(action, torch::kInt64);
  int mb = idx.numel();
  launch_ppo_gather_disc(
      idx.data_ptr<long>(), mb, obs.data_ptr<float>(), obs.size(1),
      obs_out.size(1), action.data_ptr<long>(), logp.data_ptr<float>(),
      value.data_ptr<float>(), adv.data_ptr<float>(),
      targets.data_ptr<float>(), obs_out.data_ptr(),
      action_out.data_ptr<long>(), logp_out.data_ptr<float>(),
      value_out.data_ptr<float>(), adv_out.data_ptr<float>(),
      targets_out.data_ptr<float>(), fptr_or_null(nmean), fptr_or_null(nvar),
      cur_stream());
}

void value_forward(torch::Tensor obs, torch::Tensor W1c, torch::Tensor b1c,
                   torch::Tensor W2c, torch::Tensor b2c, torch::Tensor Wvc,
                   torch::Tensor bvc, torch::Tensor value_out,
                   torch::Tensor nmean, torch::Tensor nvar) {
  CHK(obs, torch::kFloat32);
  CHK(W1c, torch::kBFloat16);
  int B = obs.size(0), OBS = obs.size(1);
  int HID = W2c.size(0);
  launch_value_forward(obs.data_ptr<float>(), W1c.data_ptr(),
                       b1c.data_ptr<float>(), W2c.data_ptr(),
                       b2c.data_ptr<float>(), Wvc.data_ptr(),
                       bvc.data_ptr<float>(), value_out.data_ptr<float>(),
                       fptr_or_null(nmean), fptr_or_null(nvar), B, OBS, HID,
                       cur_stream());
}

void rollout_step_ant(torch::Tensor obs_io, torch::Tensor env_state,
                      torch::Tensor step_count, torch::Tensor ep_return,
                      torch::Tensor ep_length, torch::Tensor last_ep_return,
                      torch::Tensor last_ep_length, torch::Tensor W1a,
                      torch::Tensor b1a, torch::Tensor W2a, torch::Tensor b2a,
                      torch::Tensor Wha, torch::Tensor bha, torch::Tensor W1c,
                      torch::Tensor b1c, torch::Tensor W2c, torch::Tensor b2c,
                      torch::Tensor Wvc, torch::Tensor bvc,
                      torch::Tensor buf_obs, torch::Tensor buf_action,
                      torch::Tensor buf_logp, torch::Tensor buf_value,
                      torch::Tensor buf_bootstrap, torch::Tensor buf_reward,
                      torch::Tensor buf_discount, torch::Tensor buf_steptype,
                      int64_t max_episode_steps, double min_scale,
                      double aff_scale, double aff_shift,
                      double log_aff_scale, int64_t policy_seed,
                      int64_t env_seed, torch::Tensor policy_draw,
                      torch::Tensor env_draw, int64_t draw_offset) {
  CHK(obs_io, torch::kFloat32);
  CHK(env_state, torch::kFloat32);
  CHK(W1a, torch::kBFloat16);
  int B = obs_io.size(0), OBS = obs_io.size(1);
  int HID = W2a.size(0), ACT = buf_action.size(1);
  launch_rollout_step_ant(
      obs_io.data_ptr<float>(), env_state.data_ptr<float>(),
      step_count.data_ptr<int>(), ep_return.data_ptr<float>(),
      ep_length.data_ptr<int>(), last_ep_return.data_ptr<float>(),
      last_ep_length.data_ptr<int>(), W1a.data_ptr(), b1a.data_ptr<float>(),
      W2a.data_ptr(), b2a.data_ptr<float>(), Wha.data_ptr(),
      bha.data_ptr<float>(), W1c.data_ptr(), b1c.data_ptr<float>(),
      W2c.data_ptr(), b2c.data_ptr<float>(), Wvc.data_ptr(),
      bvc.data_ptr<float>(), buf_obs.data_ptr<float>(),
      buf_action.data_ptr<float>(), buf_logp.data_ptr<float>(),
      buf_value.data_ptr<float>(), buf_bootstrap.data_ptr<float>(),
      buf_reward.data_ptr<float>(), buf_discount.data_ptr<float>(),
      buf_steptype.data_ptr<unsigned char>(), B, OBS, ACT, HID,
      (int)max_episode_steps, (float)min_scale, (float)aff_scale,
      (float)aff_shift, (float)log_aff_scale, (uint64_t)policy_seed,
      (uint64_t)env_seed, (unsigned int*)policy_draw.data_ptr<int>(),
      (unsigned int*)env_draw.data_ptr<int>(), (unsigned int)draw_offset,
      cur_stream());
}

void linear_silu(torch::Tensor X, torch::Tensor W, torch::Tensor bias,
                 torch::Tensor Z, torch::Tensor H, int64_t do_silu) {
  CHK(X, torch::kBFloat16);
  CHK(W, torch::kBFloat16);
  CHK(bias, torch::kFloat32);
  int S = X.size(0), K = X.size(1), N = W.size(0);
  TORCH_CHECK(S % 128 == 0 && N % 128 == 0 && K % 32 == 0,
              "linear_silu tile constraints");
  void* h = H.numel() > 0 ? H.data_ptr() : nullptr;
  launch_linear_silu(X.data_ptr(), W.data_ptr(), bias.data_ptr<float>(),
                     Z.data_ptr(), h, S, K, N, (int)do_silu, cur_stream());
}

void silu_fwd(torch::Tensor z, torch::Tensor h) {
  CHK(z, torch::kBFloat16);
  TORCH_CHECK(z.numel() % 8 == 0, "silu_fwd needs numel % 8 == 0");
  launch_silu_fwd(z.data_ptr(), h.data_ptr(), z.numel(), cur_stream());
}

void silu_bwd(torch::Tensor dh, torch::Tensor z, torch::Tensor dz) {
  CHK(z, torch::kBFloat16);
  TORCH_CHECK(z.numel() % 8 == 0, "silu_bwd needs numel % 8 == 0");
  launch_silu_bwd(dh.data_ptr(), z.data_ptr(), dz.data_ptr(), z.numel(),
                  cur_stream());
}

void ppo_gather(torch::Tensor idx, torch::Tensor obs, torch::Tensor action,
                torch::Tensor logp, torch::Tensor value, torch::Tensor adv,
                torch::Tensor targets, torch::Tensor obs_out,
                torch::Tensor action_out, torch::Tensor logp_out,
                torch::Tensor value_out, torch::Tensor adv_out,
                torch::Tensor targets_out, torch::Tensor nmean,
                torch::Tensor nvar) {
  CHK(idx, torch::kInt64);
  CHK(obs, torch::kFloat32);
  CHK(obs_out, torch::kBFloat16);
  int mb = idx.numel();
  launch_ppo_gather(idx.data_ptr<long>(), mb, obs.data_ptr<float>(),
                    obs.size(1), obs_out.size(1), action.data_ptr<float>(),
                    action.size(1),
                    logp.data_ptr<float>(), value.data_ptr<float>(),
                    adv.data_ptr<float>(), targets.data_ptr<float>(),
                    obs_out.data_ptr(), action_out.data_ptr<float>(),
                    logp_out.data_ptr<float>(), value_out.data_ptr<float>(),
                    adv_out.data_ptr<float>(), targets_out.data_ptr<float>(),
                    fptr_or_null(nmean), fptr_or_null(nvar), cur_stream());
}

void ppo_head_loss(torch::Tensor heads, torch::Tensor v_in,
                   torch::Tensor action, torch::Tensor old_logp,
                   torch::Tensor old_value, torch::Tensor adv,
                   torch::Tensor targets, torch::Tensor dhead,
                   torch::Tensor dv, torch::Tensor dv16,
                   torch::Tensor metrics, double clip_eps,
                   double ent_coef, double vf_coef, double min_scale,
                   double aff_scale, double aff_shift, double log_aff_scale,
                   int64_t seed, torch::Tensor draw_buf, int64_t draw_offset,
                   int64_t do_bump) {
  CHK(heads, torch::kBFloat16);
  CHK(v_in, torch::kBFloat16);
  CHK(action, torch::kFloat32);
  int B = heads.size(0), ACT = action.size(1);
  TORCH_CHECK(heads.size(1) == 16, "heads must be [B,16] (loc|scale packed)");
  unsigned int* db = draw_buf.numel() > 0
                         ? (unsigned int*)draw_buf.data_ptr<int>()
                         : nullptr;
  void* dv16p = dv16.numel() > 0 ? dv16.data_ptr() : nullptr;
  float* mp = metrics.numel() > 0 ? metrics.data_ptr<float>() : nullptr;
  launch_ppo_head_loss(
      heads.data_ptr(), v_in.data_ptr(), action.data_ptr<float>(),
      old_logp.data_ptr<float>(), old_value.data_ptr<float>(),
      adv.data_ptr<float>(), targets.data_ptr<float>(), dhead.data_ptr(),
      dv.data_ptr(), dv16p, mp, B, ACT,
      (float)clip_eps,
      (float)ent_coef, (float)vf_coef, (float)min_scale, (float)aff_scale,
      (float)aff_shift, (float)log_aff_scale, (uint64_t)seed, db,
      (unsigned int)draw_offset, (int)do_bump, cur_stream());
}

void wgrad(torch::Tensor dZ, torch::Tensor X, torch::Tensor slab,
           int64_t dW_off, int64_t db_off, int64_t n_valid,
           int64_t kpg = 0, int64_t ntb = 0) {
  CHK(dZ, torch::kBFloat16);
  CHK(X, torch::kBFloat16);
  CHK(slab, torch::kFloat32);
  int S = dZ.size(0), N_STRIDE = dZ.size(1), K = X.size(1);
  TORCH_CHECK(X.size(0) == S, "wgrad: dZ/X row mismatch");
  TORCH_CHECK(S % (4 * 32) == 0, "wgrad: S must be a multiple of 128");
  long stride = slab.size(1);
  launch_wgrad(dZ.data_ptr(), X.data_ptr(), slab.data_ptr<float>(),
               (long)dW_off, (long)db_off, stride, S, N_STRIDE, K,
               (int)n_valid, (int)kpg, (int)ntb, cur_stream());
}

void slab_reduce(torch::Tensor slab, torch::Tensor grad16,
                 torch::Tensor sqnorm, torch::Tensor step_t) {
  CHK(slab, torch::kFloat32);
  CHK(grad16, torch::kBFloat16);
  float* sq = sqnorm.numel() > 0 ? sqnorm.data_ptr<float>() : nullptr;
  long* st = step_t.numel() > 0 ? step_t.data_ptr<long>() : nullptr;
  launch_slab_reduce(slab.data_ptr<float>(), grad16.data_ptr(),
                     slab.size(1), grad16.numel(), sq, st, cur_stream());
}

void bump_add(torch::Tensor draw_buf, int64_t n) {
  launch_bump_add((unsigned int*)draw_buf.data_ptr<int>(), (unsigned int)n,
                  cur_stream());
}

void tr16_probe(torch::Tensor in, torch::Tensor out, int64_t base_mode) {
  CHK(in, torch::kBFloat16);
  launch_tr16_probe(in.data_ptr(), out.data_ptr<float>(), (int)base_mode,
                    cur_stream());
}

}  // namespace

void sumtree_update(torch::Tensor tree, torch::Tensor idx,
                    torch::Tensor prio, int64_t cap, int64_t depth) {
  CHK(tree, torch::kFloat32);
  CHK(idx, torch::kInt64);
  CHK(prio, torch::kFloat32);
  launch_sumtree_update(tree.data_ptr<float>(), idx.data_ptr<long>(),
                        prio.data_ptr<float>(), idx.numel(), (int)cap,
                        (int)depth, cur_stream());
}

void sumtree_sample(torch::Tensor tree, torch::Tensor u, torch::Tensor out,
                    int64_t cap, int64_t depth, int64_t n_items) {
  CHK(tree, torch::kFloat32);
  CHK(u, torch::kFloat32);
  CHK(out, torch::kInt64);
  launch_sumtree_sample(tree.data_ptr<float>(), u.data_ptr<float>(),
                        out.data_ptr<long>(), u.numel(), (int)cap, (int)depth,
                        (int)n_items, cur_stream());
}

void rnn_scan(torch::Tensor Xp, torch::Tensor Whh, torch::Tensor bhh,
              torch::Tensor resets, torch::Tensor h0, torch::Tensor c0,
              torch::Tensor Hout, torch::Tensor hT, torch::Tensor cT,
              int64_t lstm) {
  CHK(Xp, torch::kFloat32);
  CHK(Whh, torch::kBFloat16);
  CHK(resets, torch::kUInt8);
  CHK(h0, torch::kFloat32);
  int T = Xp.size(0), B = Xp.size(1);
  int H = h0.size(1);
  TORCH_CHECK(H == 128 || H == 256, "rnn_scan supports H 128/256");
  TORCH_CHECK(Xp.size(2) == (lstm ? 4 : 3) * H, "Xp gate dim mismatch");
  launch_rnn_scan(
      Xp.data_ptr<float>(), Whh.data_ptr(), bhh.data_ptr<float>(),
      resets.data_ptr<unsigned char>(), h0.data_ptr<float>(),
      c0.numel() > 0 ? c0.data_ptr<float>() : nullptr,
      Hout.data_ptr<float>(), hT.data_ptr<float>(),
      cT.numel() > 0 ? cT.data_ptr<float>() : nullptr, T, B, H, (int)lstm,
      cur_stream());
}

void snake_step(torch::Tensor grid, torch::Tensor head_r,
                torch::Tensor head_c, torch::Tensor fruit_r,
                torch::Tensor fruit_c, torch::Tensor length,
                torch::Tensor action, torch::Tensor step_count,
                torch::Tensor ep_return, torch::Tensor ep_length,
                torch::Tensor last_ep_return, torch::Tensor last_ep_length,
                torch::Tensor obs_out, torch::Tensor next_obs_out,
                torch::Tensor reward_out, torch::Tensor discount_out,
                torch::Tensor steptype_out, torch::Tensor done_out,
                int64_t max_episode_steps, int64_t seed,
                torch::Tensor draw_buf, int64_t draw_offset,
                int64_t do_bump) {
  CHK(grid, torch::kInt32);
  CHK(head_r, torch::kInt64);
  CHK(action, torch::kInt64);
  CHK(length, torch::kInt32);
  int B = grid.size(0);
  unsigned int* db = draw_buf.numel() > 0
                         ? (unsigned int*)draw_buf.data_ptr<int>()
                         : nullptr;
  launch_snake_step(
      grid.data_ptr<int>(), head_r.data_ptr<long>(), head_c.data_ptr<long>(),
      fruit_r.data_ptr<long>(), fruit_c.data_ptr<long>(),
      length.data_ptr<int>(), action.data_ptr<long>(),
      step_count.data_ptr<int>(), ep_return.data_ptr<float>(),
      ep_length.data_ptr<int>(), last_ep_return.data_ptr<float>(),
      last_ep_length.data_ptr<int>(), obs_out.data_ptr<float>(),
      next_obs_out.data_ptr<float>(), reward_out.data_ptr<float>(),
      discount_out.data_ptr<float>(),
      steptype_out.data_ptr<unsigned char>(),
      done_out.data_ptr<unsigned char>(), B, (int)max_episode_steps,
      (uint64_t)seed, db, (unsigned int)draw_offset, (int)do_bump,
      cur_stream());
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("snake_step", &snake_step,
        "fused Snake env step (dynamics + metrics + autoreset + render)");
  m.def("rnn_scan", &rnn_scan,
        "fused done-masked GRU/LSTM sequence scan (K13)");
  m.def("policy_value_step_disc", &policy_value_step_disc,
        "fused actor+critic fwd, categorical Gumbel-max sample (MFMA)");
  m.def("ppo_head_loss_disc", &ppo_head_loss_disc,
        "categorical PPO losses + analytic logits backward");
  m.def("ppo_gather_disc", &ppo_gather_disc,
        "fused minibatch gather (int64 actions)");
  m.def("sumtree_update", &sumtree_update,
        "scatter leaf priorities + repair ancestors (device sum-tree)");
  m.def("sumtree_sample", &sumtree_sample,
        "stratified proportional sum-tree descent");
  m.def("cartpole_step", &cartpole_step, "fused CartPole env step");
  m.def("ant_step", &ant_step, "fused Ant env step");
  m.def("ant_reset", &ant_reset, "Ant reset");
  m.def("humanoid_step", &humanoid_step, "fused Humanoid env step");
  m.def("gae", &gae, "truncation-aware GAE reverse scan");
  m.def("lambda_returns", &lambda_returns, "lambda returns reverse scan");
  m.def("vtrace", &vtrace, "vtrace errors + pg advantage");
  m.def("offpolicy_returns", &offpolicy_returns, "retrace-style returns");
  m.def("fused_adam", &fused_adam, "fused global-norm-clip + Adam");
  m.def("fused_adam_bf16", &fused_adam_bf16,
        "fused clip + Adam, bf16 grads + bf16 param mirror");
  m.def("polyak", &polyak, "polyak target update");
  m.def("mfma_probe", &mfma_probe, "MFMA 16x16x32 bf16 layout probe");
  m.def("policy_value_step", &policy_value_step,
        "fused actor+critic fwd + tanh-normal sample (MFMA)");
  m.def("value_forward", &value_forward, "fused critic fwd (MFMA)");
  m.def("rollout_step_ant", &rollout_step_ant,
        "fused rollout step: policy + Ant physics + bootstrap, one launch");
  m.def("linear_silu", &linear_silu,
        "fused Linear(+bias)+SiLU forward, MFMA tiled");
  m.def("silu_fwd", &silu_fwd, "bf16 silu forward");
  m.def("silu_bwd", &silu_bwd, "bf16 silu backward");
  m.def("ppo_gather", &ppo_gather, "fused minibatch gather");
  m.def("ppo_head_loss", &ppo_head_loss,
        "fused PPO head fwd + losses + analytic head bwd");
  m.def("wgrad", &wgrad, "split-K MFMA weight grad + bias colsum -> slab",
        py::arg("dZ"), py::arg("X"), py::arg("slab"), py::arg("dW_off"),
        py::arg("db_off"), py::arg("n_valid"), py::arg("kpg") = 0,
        py::arg("ntb") = 0);
  m.def("slab_reduce", &slab_reduce,
        "sum wgrad slabs into flat bf16 grads (+ fused Adam prologue)");
  m.def("bump_add", &bump_add, "add N to a device RNG draw counter");
  m.def("tr16_probe", &tr16_probe, "ds_read_tr16_b64 semantics probe");
}
