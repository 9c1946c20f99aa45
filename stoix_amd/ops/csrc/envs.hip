// Fused environment-step kernels for gfx950.
//
// Each kernel is the MI355X replacement for K1 of SURVEY.md §2.9: one launch
// does dynamics + termination + truncation + episode-metric update +
// autoreset for all B envs (the reference gets this fusion from XLA's
// compilation of scan(env.step); here it is explicit). One thread per env:
// the whole state row lives in registers; reset noise comes from Philox
// substreams keyed by (seed, env, step_counter) so replayed hip graphs stay
// deterministic per seed.
#include "common.h"
#include "ant_core.h"

// ---------------------------------------------------------------- CartPole
// Mirrors stoix_amd/envs/classic.py::CartPole (standard CartPole-v1
// dynamics) + the StatefulVecEnv wrapper semantics (envs/env.py::step).
extern "C" __global__ void cartpole_step_kernel(
    float* __restrict__ state,          // [B, 4]
    const long* __restrict__ action,    // [B]
    int* __restrict__ step_count,       // [B]
    float* __restrict__ ep_return,      // [B]
    int* __restrict__ ep_length,        // [B]
    float* __restrict__ last_ep_return, // [B]
    int* __restrict__ last_ep_length,   // [B]
    float* __restrict__ obs_out,        // [B, 4] post-autoreset obs
    float* __restrict__ next_obs_out,   // [B, 4] true final obs
    float* __restrict__ reward_out,     // [B]
    float* __restrict__ discount_out,   // [B]
    unsigned char* __restrict__ steptype_out, // [B]
    unsigned char* __restrict__ done_out,     // [B] is_terminal_step
    int B, int max_episode_steps, uint64_t seed,
    const unsigned int* __restrict__ draw_buf, unsigned int draw_offset) {
  int b = blockIdx.x * blockDim.x + threadIdx.x;
  if (b >= B) return;
  uint32_t draw = *draw_buf + draw_offset;

  float x = state[b * 4 + 0];
  float x_dot = state[b * 4 + 1];
  float th = state[b * 4 + 2];
  float th_dot = state[b * 4 + 3];

  const float GRAV = 9.8f, MC = 1.0f, MP = 0.1f, LEN = 0.5f, FMAG = 10.0f, TAU = 0.02f;
  const float THL = 12.0f * 2.0f * 3.14159265358979f / 360.0f, XL = 2.4f;

  float force = (action[b] > 0) ? FMAG : -FMAG;
  float costh = cosf(th), sinth = sinf(th);
  float total_mass = MC + MP;
  float pml = MP * LEN;
  float temp = (force + pml * th_dot * th_dot * sinth) / total_mass;
  float thacc = (GRAV * sinth - costh * temp) /
                (LEN * (4.0f / 3.0f - MP * costh * costh / total_mass));
  float xacc = temp - pml * thacc * costh / total_mass;
  x += TAU * x_dot;
  x_dot += TAU * xacc;
  th += TAU * th_dot;
  th_dot += TAU * thacc;

  bool terminated = (fabsf(x) > XL) || (fabsf(th) > THL);
  float reward = 1.0f;

  int sc = step_count[b] + 1;
  bool truncated = (sc >= max_episode_steps) && !terminated;
  bool done = terminated || truncated;

  float ret = ep_return[b] + reward;
  int len = ep_length[b] + 1;
  if (done) { last_ep_return[b] = ret; last_ep_length[b] = len; }

  // true final obs
  next_obs_out[b * 4 + 0] = x;
  next_obs_out[b * 4 + 1] = x_dot;
  next_obs_out[b * 4 + 2] = th;
  next_obs_out[b * 4 + 3] = th_dot;

  if (done) {
    Rng4 r = philox_uniform4(seed, 0u, (uint32_t)b, draw);
    x = -0.05f + 0.1f * r.a;
    x_dot = -0.05f + 0.1f * r.b;
    th = -0.05f + 0.1f * r.c;
    th_dot = -0.05f + 0.1f * r.d;
    sc = 0; ret = 0.0f; len = 0;
  }
  state[b * 4 + 0] = x;
  state[b * 4 + 1] = x_dot;
  state[b * 4 + 2] = th;
  state[b * 4 + 3] = th_dot;
  obs_out[b * 4 + 0] = x;
  obs_out[b * 4 + 1] = x_dot;
  obs_out[b * 4 + 2] = th;
  obs_out[b * 4 + 3] = th_dot;
  step_count[b] = sc;
  ep_return[b] = ret;
  ep_length[b] = len;
  reward_out[b] = reward;
  discount_out[b] = terminated ? 0.0f : 1.0f;
  steptype_out[b] = terminated ? ST_TERMINATED : (truncated ? ST_TRUNCATED : ST_MID);
  done_out[b] = done ? 1 : 0;
}

// --------------------------------------------------------------------- Ant
// Mirrors stoix_amd/envs/ant.py (Ant-class quadruped, 4 substeps of
// semi-implicit Euler with penalty contacts). One thread per env; the 29
// state floats, 8 action floats and all leg intermediates live in registers.

extern "C" __global__ void ant_step_kernel(
    float* __restrict__ state,          // [B, 29]
    const float* __restrict__ action,   // [B, 8]
    int* __restrict__ step_count,
    float* __restrict__ ep_return,
    int* __restrict__ ep_length,
    float* __restrict__ last_ep_return,
    int* __restrict__ last_ep_length,
    float* __restrict__ obs_out,        // [B, 27]
    float* __restrict__ next_obs_out,   // [B, 27]
    float* __restrict__ reward_out,
    float* __restrict__ discount_out,
    unsigned char* __restrict__ steptype_out,
    unsigned char* __restrict__ done_out,
    int B, int max_episode_steps, uint64_t seed,
    const unsigned int* __restrict__ draw_buf, unsigned int draw_offset) {
  // FOUR lanes per env (lane quad, one leg each; ant_core.h x4 physics)
  int tid = blockIdx.x * blockDim.x + threadIdx.x;
  int b = tid >> 2;
  int leg = tid & 3;
  if (b >= B) return;
  uint32_t draw = *draw_buf + draw_offset;

  float s[ANT_STATE];
#pragma unroll
  for (int i = 0; i < ANT_STATE; ++i) s[i] = state[b * ANT_STATE + i];
  float a[ANT_ACT];
#pragma unroll
  for (int i = 0; i < ANT_ACT; ++i)
    a[i] = fminf(fmaxf(action[b * ANT_ACT + i], -1.0f), 1.0f);

  float reward;
  bool terminated;
  ant_physics_step_x4(s, a, leg, &reward, &terminated);
  if (leg != 0) return;  // lane 0 of each quad does the bookkeeping

  // ---- wrapper semantics (autoreset + episode metrics)
  int sc = step_count[b] + 1;
  bool truncated = (sc >= max_episode_steps) && !terminated;
  bool done = terminated || truncated;
  float ret = ep_return[b] + reward;
  int len = ep_length[b] + 1;
  if (done) { last_ep_return[b] = ret; last_ep_length[b] = len; }

  ant_write_obs(s, next_obs_out + b * ANT_OBS);  // true final obs

  if (done) {
    ant_reset_state(s, seed, (uint32_t)b, draw);
    sc = 0; ret = 0.0f; len = 0;
  }

#pragma unroll
  for (int i = 0; i < ANT_STATE; ++i) state[b * ANT_STATE + i] = s[i];
  ant_write_obs(s, obs_out + b * ANT_OBS);
  step_count[b] = sc;
  ep_return[b] = ret;
  ep_length[b] = len;
  reward_out[b] = reward;
  discount_out[b] = terminated ? 0.0f : 1.0f;
  steptype_out[b] = terminated ? ST_TERMINATED : (truncated ? ST_TRUNCATED : ST_MID);
  done_out[b] = done ? 1 : 0;
}

// ----------------------------------------------------- humanoid (biped)
// Mirrors stoix_amd/envs/humanoid.py EXACTLY (17 damped joints with soft
// limits; two feet from hip-pitch/roll + knee; penalty contacts; forward
// reward). One env per thread; the 47 state floats live in registers.

#define HUM_STATE 47
#define HUM_OBS 45
#define HUM_ACT 17

extern "C" __global__ void humanoid_step_kernel(
    float* __restrict__ state,          // [B, 47]
    const float* __restrict__ action,   // [B, 17]
    int* __restrict__ step_count,
    float* __restrict__ ep_return,
    int* __restrict__ ep_length,
    float* __restrict__ last_ep_return,
    int* __restrict__ last_ep_length,
    float* __restrict__ obs_out,        // [B, 45]
    float* __restrict__ next_obs_out,   // [B, 45]
    float* __restrict__ reward_out,
    float* __restrict__ discount_out,
    unsigned char* __restrict__ steptype_out,
    unsigned char* __restrict__ done_out,
    int B, int max_episode_steps, uint64_t seed,
    const unsigned int* __restrict__ draw_buf, unsigned int draw_offset) {
  int b = blockIdx.x * blockDim.x + threadIdx.x;
  if (b >= B) return;
  uint32_t draw = *draw_buf + draw_offset;

  // constants (keep EXACTLY in sync with stoix_amd/envs/humanoid.py)
  const float TORSO_MASS = 40.0f, TORSO_INERTIA = 2.0f, TORSO_Z0 = 1.3f;
  const float HIP_SEP = 0.12f, LU = 0.45f, LL = 0.45f;
  const float JOINT_INERTIA = 0.12f, JOINT_DAMPING = 2.0f, GEAR = 60.0f;
  const float LIMIT = 1.2f, KNEE_LO = 0.02f, KNEE_HI = 2.0f, LIMIT_K = 80.0f;
  const float KN = 1.2e4f, KD = 300.0f, FRICTION = 1.0f, GRAV = -9.81f;
  const float DT = 0.015f;
  const int SUBSTEPS = 4;
  const float CTRL_COST = 0.1f, HEALTHY = 5.0f, FORWARD_W = 1.25f;
  const float Z_MIN = 0.8f, Z_MAX = 2.1f;
  const int R_HIP_X = 3, R_HIP_Y = 5, R_KNEE = 6;
  const int L_HIP_X = 7, L_HIP_Y = 9, L_KNEE = 10;

  float s[HUM_STATE];
#pragma unroll
  for (int i = 0; i < HUM_STATE; ++i) s[i] = state[b * HUM_STATE + i];
  float a[HUM_ACT];
#pragma unroll
  for (int i = 0; i < HUM_ACT; ++i)
    a[i] = fminf(fmaxf(action[b * HUM_ACT + i], -1.0f), 1.0f);

  float x_before = s[0];
  const float dt = DT / SUBSTEPS;

  for (int sub = 0; sub < SUBSTEPS; ++sub) {
    float* pos = s + 0;
    float* quat = s + 3;
    float* linvel = s + 7;
    float* angvel = s + 10;
    float* qpos = s + 13;   // 17
    float* qvel = s + 30;   // 17

    // ---- joints first (humanoid.py updates qpos before the foot calc)
#pragma unroll
    for (int j = 0; j < HUM_ACT; ++j) {
      float lo = -LIMIT, hi = LIMIT;
      if (j == R_KNEE || j == L_KNEE) { lo = KNEE_LO; hi = KNEE_HI; }
      float limit_tau = -LIMIT_K * (fmaxf(qpos[j] - hi, 0.0f) - fmaxf(lo - qpos[j], 0.0f));
      float qacc = (GEAR * a[j] - JOINT_DAMPING * qvel[j] + limit_tau) / JOINT_INERTIA;
      qvel[j] += dt * qacc;
      qpos[j] += dt * qvel[j];
    }

    // ---- feet from NEW joint angles, torso pose still old
    V3 total_f = {0.f, 0.f, 0.f}, total_tau = {0.f, 0.f, 0.f};
#pragma unroll
    for (int side = 0; side < 2; ++side) {
      int hx = side == 0 ? R_HIP_X : L_HIP_X;
      int hy = side == 0 ? R_HIP_Y : L_HIP_Y;
      int kn = side == 0 ? R_KNEE : L_KNEE;
      float hp = qpos[hy], hr = qpos[hx], k = qpos[kn];
      V3 body_off = {LU * sinf(hp) + LL * sinf(hp + k),
                     (LU + LL) * sinf(hr) + (side == 0 ? HIP_SEP : -HIP_SEP),
                     -(LU * cosf(hp) + LL * cosf(hp + k))};
      V3 r = quat_rot(quat, body_off);
      V3 foot_w = {pos[0] + r.x, pos[1] + r.y, pos[2] + r.z};
      V3 av = {angvel[0], angvel[1], angvel[2]};
      V3 fv = add3(v3(linvel[0], linvel[1], linvel[2]), cross3(av, r));
      float pen = fmaxf(-foot_w.z, 0.0f);
      float fn = fmaxf(KN * pen - KD * fv.z, 0.0f);
      if (pen <= 0.0f) fn = 0.0f;
      V3 cf = {-FRICTION * fn * tanhf(4.0f * fv.x),
               -FRICTION * fn * tanhf(4.0f * fv.y), fn};
      total_f = add3(total_f, cf);
      total_tau = add3(total_tau, cross3(r, cf));
    }

    // ---- torso integration
    linvel[0] += dt * (total_f.x / TORSO_MASS);
    linvel[1] += dt * (total_f.y / TORSO_MASS);
    linvel[2] += dt * (total_f.z / TORSO_MASS + GRAV);
    pos[0] += dt * linvel[0];
    pos[1] += dt * linvel[1];
    pos[2] += dt * linvel[2];
    angvel[0] += dt * (total_tau.x / TORSO_INERTIA - 0.5f * angvel[0]);
    angvel[1] += dt * (total_tau.y / TORSO_INERTIA - 0.5f * angvel[1]);
    angvel[2] += dt * (total_tau.z / TORSO_INERTIA - 0.5f * angvel[2]);
    // quaternion integration (matches ant.py quat_integrate)
    {
      float w = quat[0], x = quat[1], y = quat[2], z = quat[3];
      float ox = angvel[0], oy = angvel[1], oz = angvel[2];
      float dw = 0.5f * (-x * ox - y * oy - z * oz);
      float dx = 0.5f * (w * ox + y * oz - z * oy);
      float dy = 0.5f * (w * oy + z * ox - x * oz);
      float dz = 0.5f * (w * oz + x * oy - y * ox);
      float nw = w + dt * dw, nx = x + dt * dx, ny = y + dt * dy, nz = z + dt * dz;
      float nrm = sqrtf(fmaxf(nw * nw + nx * nx + ny * ny + nz * nz, 1e-16f));
      quat[0] = nw / nrm; quat[1] = nx / nrm; quat[2] = ny / nrm; quat[3] = nz / nrm;
    }
  }

  // ---- reward / termination
  float forward_vel = (s[0] - x_before) / DT;
  float ctrl = 0.0f;
#pragma unroll
  for (int i = 0; i < HUM_ACT; ++i) ctrl += a[i] * a[i];
  float z = s[2];
  bool finite = true;
#pragma unroll
  for (int i = 0; i < HUM_STATE; ++i) finite = finite && isfinite(s[i]);
  bool healthy = (z > Z_MIN) && (z < Z_MAX) && finite;
  float reward = FORWARD_W * forward_vel + HEALTHY - CTRL_COST * ctrl;
  bool terminated = !healthy;
  if (!finite) {
#pragma unroll
    for (int i = 0; i < HUM_STATE; ++i) s[i] = 0.0f;
    reward = isfinite(reward) ? reward : 0.0f;
  }

  int sc = step_count[b] + 1;
  bool truncated = (!terminated) && (sc >= max_episode_steps);
  bool done = terminated || truncated;
  float ret = ep_return[b] + reward;
  int len = ep_length[b] + 1;
  if (done) { last_ep_return[b] = ret; last_ep_length[b] = len; }

  // true final obs
  {
    float* o = next_obs_out + b * HUM_OBS;
    o[0] = s[2];
#pragma unroll
    for (int i = 0; i < 4; ++i) o[1 + i] = s[3 + i];
#pragma unroll
    for (int i = 0; i < 3; ++i) o[5 + i] = s[7 + i];
#pragma unroll
    for (int i = 0; i < 3; ++i) o[8 + i] = s[10 + i];
#pragma unroll
    for (int i = 0; i < 17; ++i) o[11 + i] = s[13 + i];
#pragma unroll
    for (int i = 0; i < 17; ++i) o[28 + i] = s[30 + i];
  }

  if (done) {
    // autoreset, mirroring Humanoid._reset_fn (philox stream 4)
    float u[36];
#pragma unroll
    for (int blk = 0; blk < 9; ++blk) {
      Rng4 r = philox_uniform4(seed, 4u, (uint32_t)b, draw * 16u + blk);
      u[blk * 4 + 0] = r.a; u[blk * 4 + 1] = r.b;
      u[blk * 4 + 2] = r.c; u[blk * 4 + 3] = r.d;
    }
#pragma unroll
    for (int i = 0; i < HUM_STATE; ++i) s[i] = 0.0f;
    s[2] = TORSO_Z0;
    s[3] = 1.0f;
#pragma unroll
    for (int i = 0; i < 17; ++i) s[13 + i] = -0.03f + 0.06f * u[i];
    s[13 + R_KNEE] += 0.15f;
    s[13 + L_KNEE] += 0.15f;
#pragma unroll
    for (int i = 0; i < 17; ++i) s[30 + i] = -0.02f + 0.04f * u[17 + i];
    sc = 0; ret = 0.0f; len = 0;
  }

#pragma unroll
  for (int i = 0; i < HUM_STATE; ++i) state[b * HUM_STATE + i] = s[i];
  {
    float* o = obs_out + b * HUM_OBS;
    o[0] = s[2];
#pragma unroll
    for (int i = 0; i < 4; ++i) o[1 + i] = s[3 + i];
#pragma unroll
    for (int i = 0; i < 3; ++i) o[5 + i] = s[7 + i];
#pragma unroll
    for (int i = 0; i < 3; ++i) o[8 + i] = s[10 + i];
#pragma unroll
    for (int i = 0; i < 17; ++i) o[11 + i] = s[13 + i];
#pragma unroll
    for (int i = 0; i < 17; ++i) o[28 + i] = s[30 + i];
  }
  step_count[b] = sc;
  ep_return[b] = ret;
  ep_length[b] = len;
  reward_out[b] = reward;
  discount_out[b] = terminated ? 0.0f : 1.0f;
  steptype_out[b] = terminated ? ST_TERMINATED : (truncated ? ST_TRUNCATED : ST_MID);
  done_out[b] = done ? 1 : 0;
}

// ----------------------------------------------------------- ant reset
extern "C" __global__ void ant_reset_kernel(
    float* __restrict__ state, int B, uint64_t seed, uint32_t draw) {
  int b = blockIdx.x * blockDim.x + threadIdx.x;
  if (b >= B) return;
  float s[ANT_STATE];
  ant_reset_state(s, seed, (uint32_t)b, draw);
#pragma unroll
  for (int i = 0; i < ANT_STATE; ++i) state[b * ANT_STATE + i] = s[i];
}

extern "C" __global__ void bump_u32_kernel(unsigned int* p) {
  if (threadIdx.x == 0 && blockIdx.x == 0) *p += 1;
}

// ----------------------------------------------------------- host launchers
extern "C" void launch_cartpole_step(
    float* state, const long* action, int* step_count, float* ep_return,
    int* ep_length, float* last_ep_return, int* last_ep_length, float* obs_out,
    float* next_obs_out, float* reward_out, float* discount_out,
    unsigned char* steptype_out, unsigned char* done_out, int B,
    int max_episode_steps, uint64_t seed, unsigned int* draw_buf,
    unsigned int draw_offset, int do_bump, void* stream) {
  int threads = 64;
  int blocks = (B + threads - 1) / threads;
  hipLaunchKernelGGL(cartpole_step_kernel, dim3(blocks), dim3(threads), 0,
                     (hipStream_t)stream, state, action, step_count, ep_return,
                     ep_length, last_ep_return, last_ep_length, obs_out,
                     next_obs_out, reward_out, discount_out, steptype_out,
                     done_out, B, max_episode_steps, seed, draw_buf,
                     draw_offset);
  if (do_bump)
    hipLaunchKernelGGL(bump_u32_kernel, dim3(1), dim3(1), 0,
                       (hipStream_t)stream, draw_buf);
}

extern "C" void launch_ant_step(
    float* state, const float* action, int* step_count, float* ep_return,
    int* ep_length, float* last_ep_return, int* last_ep_length, float* obs_out,
    float* next_obs_out, float* reward_out, float* discount_out,
    unsigned char* steptype_out, unsigned char* done_out, int B,
    int max_episode_steps, uint64_t seed, unsigned int* draw_buf,
    unsigned int draw_offset, int do_bump, void* stream) {
  // 4 threads per env (quad-cooperative physics), 64-thread blocks
  int threads = 64;
  int blocks = (B * 4 + threads - 1) / threads;
  hipLaunchKernelGGL(ant_step_kernel, dim3(blocks), dim3(threads), 0,
                     (hipStream_t)stream, state, action, step_count, ep_return,
                     ep_length, last_ep_return, last_ep_length, obs_out,
                     next_obs_out, reward_out, discount_out, steptype_out,
                     done_out, B, max_episode_steps, seed, draw_buf,
                     draw_offset);
  if (do_bump)
    hipLaunchKernelGGL(bump_u32_kernel, dim3(1), dim3(1), 0,
                       (hipStream_t)stream, draw_buf);
}

extern "C" void launch_humanoid_step(
    float* state, const float* action, int* step_count, float* ep_return,
    int* ep_length, float* last_ep_return, int* last_ep_length, float* obs_out,
    float* next_obs_out, float* reward_out, float* discount_out,
    unsigned char* steptype_out, unsigned char* done_out, int B,
    int max_episode_steps, uint64_t seed, unsigned int* draw_buf,
    unsigned int draw_offset, int do_bump, void* stream) {
  int threads = 64;
  int blocks = (B + threads - 1) / threads;
  hipLaunchKernelGGL(humanoid_step_kernel, dim3(blocks), dim3(threads), 0,
                     (hipStream_t)stream, state, action, step_count, ep_return,
                     ep_length, last_ep_return, last_ep_length, obs_out,
                     next_obs_out, reward_out, discount_out, steptype_out,
                     done_out, B, max_episode_steps, seed, draw_buf,
                     draw_offset);
  if (do_bump)
    hipLaunchKernelGGL(bump_u32_kernel, dim3(1), dim3(1), 0,
                       (hipStream_t)stream, draw_buf);
}

extern "C" void launch_ant_reset(float* state, int B, uint64_t seed,
                                 uint32_t draw, void* stream) {
  int threads = 256;
  int blocks = (B + threads - 1) / threads;
  hipLaunchKernelGGL(ant_reset_kernel, dim3(blocks), dim3(threads), 0,
                     (hipStream_t)stream, state, B, seed, draw);
}
