// Reverse time-scan kernels: GAE, lambda returns, vtrace (K4/K12 of
// SURVEY.md §2.9). The [T, B] recursion is sequential in T and parallel in
// B: one thread per env column walks t = T-1..0. At each t a wave's 64
// threads touch 64 consecutive elements of row t — fully coalesced. For the
// bench shapes (T=128, B=4096) the whole pass is a single ~100 us-scale
// launch replacing a 128-step Python loop.
#include "common.h"

extern "C" __global__ void gae_kernel(
    const float* __restrict__ r_t,        // [T, B]
    const float* __restrict__ discount_t, // [T, B] (gamma folded in)
    const float* __restrict__ v_tm1,      // [T, B]
    const float* __restrict__ v_t,        // [T, B] bootstrap values
    const unsigned char* __restrict__ trunc_t, // [T, B] or nullptr
    float* __restrict__ adv_out,          // [T, B]
    float* __restrict__ target_out,       // [T, B]
    int T, int B, float lambda_) {
  int b = blockIdx.x * blockDim.x + threadIdx.x;
  if (b >= B) return;
  float acc = 0.0f;
  for (int t = T - 1; t >= 0; --t) {
    int i = t * B + b;
    float d = discount_t[i];
    float delta = r_t[i] + d * v_t[i] - v_tm1[i];
    float cont = (trunc_t != nullptr && trunc_t[i]) ? 0.0f : 1.0f;
    acc = delta + d * lambda_ * cont * acc;
    adv_out[i] = acc;
    target_out[i] = acc + v_tm1[i];
  }
}

extern "C" __global__ void lambda_returns_kernel(
    const float* __restrict__ r_t,
    const float* __restrict__ discount_t,
    const float* __restrict__ v_t,
    float* __restrict__ out,
    int T, int B, float lambda_) {
  int b = blockIdx.x * blockDim.x + threadIdx.x;
  if (b >= B) return;
  float acc = v_t[(T - 1) * B + b];
  for (int t = T - 1; t >= 0; --t) {
    int i = t * B + b;
    acc = r_t[i] + discount_t[i] * ((1.0f - lambda_) * v_t[i] + lambda_ * acc);
    out[i] = acc;
  }
}

extern "C" __global__ void vtrace_kernel(
    const float* __restrict__ v_tm1,
    const float* __restrict__ v_t,
    const float* __restrict__ r_t,
    const float* __restrict__ discount_t,
    const float* __restrict__ rho_tm1,
    float* __restrict__ errors_out,
    float* __restrict__ pg_adv_out,
    int T, int B, float lambda_, float clip_rho, float clip_pg_rho) {
  int b = blockIdx.x * blockDim.x + threadIdx.x;
  if (b >= B) return;
  // backward pass for vs - v
  float acc = 0.0f;
  for (int t = T - 1; t >= 0; --t) {
    int i = t * B + b;
    float rho = rho_tm1[i];
    float rho_c = fminf(rho, clip_rho);
    float c = lambda_ * fminf(rho, 1.0f);
    float delta = rho_c * (r_t[i] + discount_t[i] * v_t[i] - v_tm1[i]);
    acc = delta + discount_t[i] * c * acc;
    errors_out[i] = acc;
  }
  // forward pass for pg advantages: q = r + d * vs_{t+1}
  for (int t = 0; t < T; ++t) {
    int i = t * B + b;
    float vs_next;
    if (t + 1 < T) {
      vs_next = errors_out[(t + 1) * B + b] + v_tm1[(t + 1) * B + b];
    } else {
      vs_next = v_t[(T - 1) * B + b];
    }
    float q = r_t[i] + discount_t[i] * vs_next;
    float pg_rho = fminf(rho_tm1[i], clip_pg_rho);
    pg_adv_out[i] = pg_rho * (q - v_tm1[i]);
  }
}

// Retrace / general off-policy corrected returns (reference multistep.py:210-311)
extern "C" __global__ void offpolicy_returns_kernel(
    const float* __restrict__ q_t,
    const float* __restrict__ v_t,
    const float* __restrict__ r_t,
    const float* __restrict__ discount_t,
    const float* __restrict__ c_t,
    float* __restrict__ out,
    int T, int B) {
  int b = blockIdx.x * blockDim.x + threadIdx.x;
  if (b >= B) return;
  int i = (T - 1) * B + b;
  float g = r_t[i] + discount_t[i] * v_t[i];
  out[i] = g;
  for (int t = T - 2; t >= 0; --t) {
    i = t * B + b;
    g = r_t[i] + discount_t[i] * (v_t[i] - c_t[i] * q_t[i] + c_t[i] * g);
    out[i] = g;
  }
}

// ----------------------------------------------------------- host launchers
extern "C" void launch_gae(const float* r_t, const float* discount_t,
                           const float* v_tm1, const float* v_t,
                           const unsigned char* trunc_t, float* adv_out,
                           float* target_out, int T, int B, float lambda_,
                           void* stream) {
  int threads = 256;
  int blocks = (B + threads - 1) / threads;
  hipLaunchKernelGGL(gae_kernel, dim3(blocks), dim3(threads), 0,
                     (hipStream_t)stream, r_t, discount_t, v_tm1, v_t, trunc_t,
                     adv_out, target_out, T, B, lambda_);
}

extern "C" void launch_lambda_returns(const float* r_t, const float* discount_t,
                                      const float* v_t, float* out, int T,
                                      int B, float lambda_, void* stream) {
  int threads = 256;
  int blocks = (B + threads - 1) / threads;
  hipLaunchKernelGGL(lambda_returns_kernel, dim3(blocks), dim3(threads), 0,
                     (hipStream_t)stream, r_t, discount_t, v_t, out, T, B,
                     lambda_);
}

extern "C" void launch_vtrace(const float* v_tm1, const float* v_t,
                              const float* r_t, const float* discount_t,
                              const float* rho_tm1, float* errors_out,
                              float* pg_adv_out, int T, int B, float lambda_,
                              float clip_rho, float clip_pg_rho, void* stream) {
  int threads = 256;
  int blocks = (B + threads - 1) / threads;
  hipLaunchKernelGGL(vtrace_kernel, dim3(blocks), dim3(threads), 0,
                     (hipStream_t)stream, v_tm1, v_t, r_t, discount_t, rho_tm1,
                     errors_out, pg_adv_out, T, B, lambda_, clip_rho,
                     clip_pg_rho);
}

extern "C" void launch_offpolicy_returns(const float* q_t, const float* v_t,
                                         const float* r_t,
                                         const float* discount_t,
                                         const float* c_t, float* out, int T,
                                         int B, void* stream) {
  int threads = 256;
  int blocks = (B + threads - 1) / threads;
  hipLaunchKernelGGL(offpolicy_returns_kernel, dim3(blocks), dim3(threads), 0,
                     (hipStream_t)stream, q_t, v_t, r_t, discount_t, c_t, out,
                     T, B);
}
