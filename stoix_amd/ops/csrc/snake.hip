// Fused Snake-v1-class env step (K1 for BASELINE config #5: Anakin
// Rainbow on Snake). Mirrors stoix_amd/envs/snake.py exactly: countdown-
// grid body, fruit eat/grow, wall/self collision, plus the
// StatefulVecEnv wrapper semantics (truncation, autoreset with the true
// final obs in next_obs, latched episode metrics) and the [12,12,5]
// observation render — ONE launch replaces the ~45 torch kernels per
// step that dominated the captured Rainbow update at small batches.
//
// One THREAD per env: the per-env work is a handful of 144-cell passes
// (decrement, fruit scan, render 720 floats) — serial per env, massively
// parallel across envs. Fruit respawn draws ONE uniform per event and
// takes the k-th empty cell (exact uniform over empties, same
// distribution as the torch path's Gumbel-max; the torch/HIP RNG streams
// differ by construction, parity tests resync on eat/reset like the
// CartPole reset-noise protocol).
#include "common.h"

namespace {
constexpr int R = 12, C = 12, NC = R * C;
constexpr int DR[4] = {-1, 0, 1, 0};
constexpr int DC[4] = {0, 1, 0, -1};
}  // namespace

extern "C" __global__ void snake_step_kernel(
    int* __restrict__ grid,        // [B, 144] countdown body grid
    long* __restrict__ head_r,     // [B]
    long* __restrict__ head_c,     // [B]
    long* __restrict__ fruit_r,    // [B]
    long* __restrict__ fruit_c,    // [B]
    int* __restrict__ length,      // [B]
    const long* __restrict__ action,  // [B]
    int* __restrict__ step_count, float* __restrict__ ep_return,
    int* __restrict__ ep_length, float* __restrict__ last_ep_return,
    int* __restrict__ last_ep_length,
    float* __restrict__ obs_out,       // [B, 12, 12, 5] post-autoreset
    float* __restrict__ next_obs_out,  // [B, 12, 12, 5] true final obs
    float* __restrict__ reward_out, float* __restrict__ discount_out,
    unsigned char* __restrict__ steptype_out,
    unsigned char* __restrict__ done_out, int B, int max_episode_steps,
    uint64_t seed, const unsigned int* __restrict__ draw_buf,
    unsigned int draw_offset) {
  int b = blockIdx.x * blockDim.x + threadIdx.x;
  if (b >= B) return;
  uint32_t draw = (draw_buf ? *draw_buf : 0u) + draw_offset;
  int* g = grid + (long)b * NC;

  int a = (int)action[b];
  if (a < 0) a = 0;
  if (a > 3) a = 3;
  int hr = (int)head_r[b], hc = (int)head_c[b];
  int nr = hr + DR[a], nc = hc + DC[a];
  bool hit_wall = nr < 0 || nr >= R || nc < 0 || nc >= C;
  int nr_s = nr < 0 ? 0 : (nr > R - 1 ? R - 1 : nr);
  int nc_s = nc < 0 ? 0 : (nc > C - 1 ? C - 1 : nc);
  bool ate = (nr_s == (int)fruit_r[b]) && (nc_s == (int)fruit_c[b]) && !hit_wall;
  int dec = ate ? 0 : 1;
#pragma unroll 4
  for (int k = 0; k < NC; ++k) {
    int v = g[k];
    if (v > 0) g[k] = v - dec;
  }
  int target = nr_s * C + nc_s;
  bool hit_self = g[target] > 0;
  bool terminated = hit_wall || hit_self;
  int len = length[b] + (ate ? 1 : 0);
  if (!terminated) g[target] = len;
  float reward = ate ? 1.0f : 0.0f;
  int fr = (int)fruit_r[b], fc = (int)fruit_c[b];
  if (ate) {
    // uniform over empty cells: count, draw, pick the k-th
    int empties = 0;
    for (int k = 0; k < NC; ++k) empties += (g[k] <= 0) ? 1 : 0;
    if (empties > 0) {
      Rng4 u = philox_uniform4(seed, 9u, (uint32_t)b, draw);
      int pick = (int)(u.a * (float)empties);
      if (pick > empties - 1) pick = empties - 1;
      int seen = 0;
      for (int k = 0; k < NC; ++k) {
        if (g[k] <= 0) {
          if (seen == pick) { fr = k / C; fc = k % C; break; }
          ++seen;
        }
      }
    }
  }
  int new_hr = terminated ? hr : nr_s;
  int new_hc = terminated ? hc : nc_s;

  // ---- wrapper semantics: truncation, metrics, autoreset
  int sc = step_count[b] + 1;
  bool truncated = (sc >= max_episode_steps) && !terminated;
  bool done = terminated || truncated;
  float ret = ep_return[b] + reward;
  int elen = ep_length[b] + 1;
  if (done) { last_ep_return[b] = ret; last_ep_length[b] = elen; }

  // ---- render the TRUE FINAL observation
  float inv_len = 1.0f / (float)(len < 1 ? 1 : len);
  float* nobs = next_obs_out + (long)b * NC * 5;
  for (int k = 0; k < NC; ++k) {
    int v = g[k];
    float* cell = nobs + k * 5;
    cell[0] = v > 0 ? 1.0f : 0.0f;                      // body
    cell[1] = (k == new_hr * C + new_hc) ? 1.0f : 0.0f;  // head
    cell[2] = v == 1 ? 1.0f : 0.0f;                     // tail
    cell[3] = (k == fr * C + fc) ? 1.0f : 0.0f;         // fruit
    cell[4] = (float)v * inv_len;                       // order
  }

  if (done) {
    // autoreset: length-1 snake at the centre, fresh fruit over empties
    for (int k = 0; k < NC; ++k) g[k] = 0;
    new_hr = R / 2;
    new_hc = C / 2;
    len = 1;
    g[new_hr * C + new_hc] = 1;
    Rng4 u = philox_uniform4(seed, 10u, (uint32_t)b, draw);
    int pick = (int)(u.b * (float)(NC - 1));
    if (pick > NC - 2) pick = NC - 2;
    // k-th empty among the 143 non-head cells
    int seen = 0;
    fr = 0; fc = 1;
    for (int k = 0; k < NC; ++k) {
      if (k == new_hr * C + new_hc) continue;
      if (seen == pick) { fr = k / C; fc = k % C; break; }
      ++seen;
    }
    sc = 0; ret = 0.0f; elen = 0;
    float* obs = obs_out + (long)b * NC * 5;
    for (int k = 0; k < NC; ++k) {
      float* cell = obs + k * 5;
      bool is_head = k == new_hr * C + new_hc;
      cell[0] = is_head ? 1.0f : 0.0f;
      cell[1] = is_head ? 1.0f : 0.0f;
      cell[2] = is_head ? 1.0f : 0.0f;  // countdown 1 == tail
      cell[3] = (k == fr * C + fc) ? 1.0f : 0.0f;
      cell[4] = is_head ? 1.0f : 0.0f;
    }
  } else {
    float* obs = obs_out + (long)b * NC * 5;
    const float* src = nobs;
    for (int k = 0; k < NC * 5; ++k) obs[k] = src[k];
  }

  head_r[b] = new_hr;
  head_c[b] = new_hc;
  fruit_r[b] = fr;
  fruit_c[b] = fc;
  length[b] = len;
  step_count[b] = sc;
  ep_return[b] = ret;
  ep_length[b] = elen;
  reward_out[b] = reward;
  discount_out[b] = terminated ? 0.0f : 1.0f;
  steptype_out[b] = terminated ? 2 : (truncated ? 3 : 1);
  done_out[b] = done ? 1 : 0;
}

extern "C" __global__ void snake_bump_kernel(unsigned int* p) {
  if (blockIdx.x == 0 && threadIdx.x == 0) (*p)++;
}

extern "C" void launch_snake_step(
    int* grid, long* head_r, long* head_c, long* fruit_r, long* fruit_c,
    int* length, const long* action, int* step_count, float* ep_return,
    int* ep_length, float* last_ep_return, int* last_ep_length,
    float* obs_out, float* next_obs_out, float* reward_out,
    float* discount_out, unsigned char* steptype_out,
    unsigned char* done_out, int B, int max_episode_steps, uint64_t seed,
    unsigned int* draw_buf, unsigned int draw_offset, int do_bump,
    void* stream) {
  int threads = 64;
  int blocks = (B + threads - 1) / threads;
  hipLaunchKernelGGL(snake_step_kernel, dim3(blocks), dim3(threads), 0,
                     (hipStream_t)stream, grid, head_r, head_c, fruit_r,
                     fruit_c, length, action, step_count, ep_return,
                     ep_length, last_ep_return, last_ep_length, obs_out,
                     next_obs_out, reward_out, discount_out, steptype_out,
                     done_out, B, max_episode_steps, seed, draw_buf,
                     draw_offset);
  if (draw_buf && do_bump)
    hipLaunchKernelGGL(snake_bump_kernel, dim3(1), dim3(1), 0,
                       (hipStream_t)stream, draw_buf);
}
