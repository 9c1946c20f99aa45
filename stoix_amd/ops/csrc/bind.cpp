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
                          void*);
void launch_ant_step(float*, const float*, int*, float*, int*, float*, int*,
                     float*, float*, float*, float*, unsigned char*,
                     unsigned char*, int, int, uint64_t, unsigned int*, void*);
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
void launch_polyak(const float*, float*, long, float, void*);
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
                   int64_t seed, torch::Tensor draw_buf) {
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
      (uint64_t)seed, (unsigned int*)draw_buf.data_ptr<int>(), cur_stream());
}

void ant_step(torch::Tensor state, torch::Tensor action,
              torch::Tensor step_count, torch::Tensor ep_return,
              torch::Tensor ep_length, torch::Tensor last_ep_return,
              torch::Tensor last_ep_length, torch::Tensor obs_out,
              torch::Tensor next_obs_out, torch::Tensor reward_out,
              torch::Tensor discount_out, torch::Tensor steptype_out,
              torch::Tensor done_out, int64_t max_episode_steps, int64_t seed,
              torch::Tensor draw_buf) {
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
      (uint64_t)seed, (unsigned int*)draw_buf.data_ptr<int>(), cur_stream());
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

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("cartpole_step", &cartpole_step, "fused CartPole env step");
  m.def("ant_step", &ant_step, "fused Ant env step");
  m.def("ant_reset", &ant_reset, "Ant reset");
  m.def("gae", &gae, "truncation-aware GAE reverse scan");
  m.def("lambda_returns", &lambda_returns, "lambda returns reverse scan");
  m.def("vtrace", &vtrace, "vtrace errors + pg advantage");
  m.def("offpolicy_returns", &offpolicy_returns, "retrace-style returns");
  m.def("fused_adam", &fused_adam, "fused global-norm-clip + Adam");
  m.def("polyak", &polyak, "polyak target update");
}
