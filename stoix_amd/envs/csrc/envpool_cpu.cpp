// Native batched CPU environment engine (envpool-equivalent).
//
// The reference's envpool suite is a C++ batched env pool feeding Sebulba's
// CPU actor threads (/root/reference/stoix/utils/make_env.py envpool rows;
// SURVEY.md §2.2/§8.6). This is the MI355X-framework counterpart: the whole
// per-step pipeline — physics, termination/truncation, episode metrics,
// autoreset, and FRAME RENDERING — runs fused in one C++ call, parallelised
// over envs with at::parallel_for. The torch-ops Breakout
// (stoix_amd/envs/breakout.py) renders via an einsum over 72 brick masks
// (~0.5 MFLOP/env/step plus ~30 kernel dispatches); this path draws the
// ~1.1k changed pixels directly (~10 us/env single-thread), which is what
// makes CPU actors keep a GPU learner fed (BASELINE config #4).
//
// Games: Breakout (rules mirror stoix_amd/envs/breakout.py exactly:
// 84x84 grayscale, 4 actions, +1/brick, terminate on miss or clear) and
// Pong (scripted tracking opponent, +-1 per point, first to 21).
#include <torch/extension.h>
#include <ATen/Parallel.h>

#include <cmath>
#include <cstdint>

namespace {

constexpr int H = 84, W = 84;
constexpr int BRICK_ROWS = 6, BRICK_COLS = 12;
constexpr int BRICK_W = W / BRICK_COLS, BRICK_H = 3;
constexpr int BRICK_TOP = 12;
constexpr int PADDLE_W = 12, PADDLE_Y = 80;
constexpr float PADDLE_SPEED = 3.0f, BALL_SPEED = 1.8f;
constexpr int NBRICK = BRICK_ROWS * BRICK_COLS;
// state row: paddle_x, ball_x, ball_y, ball_vx, ball_vy, pad, bricks[72]
constexpr int SOFF = 6;
constexpr int SDIM = SOFF + NBRICK;

constexpr uint8_t ST_MID = 1, ST_TERMINATED = 2, ST_TRUNCATED = 3;

// counter-based RNG (splitmix64) -> uniform [0,1)
inline float hash_uniform(uint64_t seed, uint64_t a, uint64_t b) {
  uint64_t x = seed + 0x9e3779b97f4a7c15ull * (a + 1) + 0xbf58476d1ce4e5b9ull * (b + 1);
  x ^= x >> 30; x *= 0xbf58476d1ce4e5b9ull;
  x ^= x >> 27; x *= 0x94d049bb133111ebull;
  x ^= x >> 31;
  return (float)(x >> 40) * (1.0f / 16777216.0f);
}

inline void reset_env(float* s, uint64_t seed, uint64_t env, uint64_t draw) {
  s[0] = W / 2.0f;
  s[1] = W * 0.3f + hash_uniform(seed, env * 2, draw) * (W * 0.4f);
  s[2] = 46.0f;
  s[3] = (hash_uniform(seed, env * 2 + 1, draw) > 0.5f) ? BALL_SPEED * 0.7f
                                                        : -BALL_SPEED * 0.7f;
  s[4] = BALL_SPEED;
  s[5] = 0.0f;
  for (int k = 0; k < NBRICK; ++k) s[SOFF + k] = 1.0f;
}

inline void render(const float* s, float* obs) {
  std::memset(obs, 0, sizeof(float) * H * W);
  // bricks (value 0.6; each brick leaves a 1px gutter right/bottom)
  for (int r = 0; r < BRICK_ROWS; ++r) {
    int y0 = BRICK_TOP + r * BRICK_H;
    for (int c = 0; c < BRICK_COLS; ++c) {
      if (s[SOFF + r * BRICK_COLS + c] <= 0.0f) continue;
      int x0 = c * BRICK_W;
      for (int y = y0; y < y0 + BRICK_H - 1; ++y)
        for (int x = x0; x < x0 + BRICK_W - 1; ++x) obs[y * W + x] = 0.6f;
    }
  }
  // paddle (2 rows)
  int px = (int)s[0];
  if (px < PADDLE_W / 2) px = PADDLE_W / 2;
  if (px > W - 1 - PADDLE_W / 2) px = W - 1 - PADDLE_W / 2;
  for (int dx = -(PADDLE_W / 2); dx < PADDLE_W / 2; ++dx) {
    int x = px + dx;
    if (x < 0) x = 0;
    if (x > W - 1) x = W - 1;
    obs[PADDLE_Y * W + x] = 1.0f;
    obs[(PADDLE_Y + 1) * W + x] = 1.0f;
  }
  // ball (2x2)
  int by = (int)s[2], bx = (int)s[1];
  if (by < 0) by = 0;
  if (by > H - 2) by = H - 2;
  if (bx < 0) bx = 0;
  if (bx > W - 2) bx = W - 2;
  for (int dy = 0; dy < 2; ++dy)
    for (int dx = 0; dx < 2; ++dx) obs[(by + dy) * W + bx + dx] = 1.0f;
}

}  // namespace

void breakout_cpu_reset(torch::Tensor state, torch::Tensor obs,
                        int64_t seed, int64_t draw) {
  TORCH_CHECK(state.size(1) == SDIM, "state must be [B, ", SDIM, "]");
  int64_t B = state.size(0);
  float* sp = state.data_ptr<float>();
  float* op = obs.data_ptr<float>();
  // serial per-call stepping (envpool design): parallelism comes from the
  // ACTOR THREADS, each stepping its own pool with the GIL released — the
  // shared intra-op pool (at::parallel_for) collapses on big hosts (256
  // cores: 23.8K SPS vs 353K serial, and worker-thread OMP regions spawn
  // per-caller teams)
  auto serial_loop = [&](int64_t lo, int64_t hi) {
    for (int64_t b = lo; b < hi; ++b) {
      reset_env(sp + b * SDIM, (uint64_t)seed, (uint64_t)b, (uint64_t)draw);
      render(sp + b * SDIM, op + b * H * W);
    }
  };
  serial_loop(0, B);
}

void breakout_cpu_step(torch::Tensor state, torch::Tensor action,
                       torch::Tensor step_count, torch::Tensor ep_return,
                       torch::Tensor ep_length, torch::Tensor last_ep_return,
                       torch::Tensor last_ep_length, torch::Tensor obs,
                       torch::Tensor next_obs, torch::Tensor reward,
                       torch::Tensor discount, torch::Tensor steptype,
                       torch::Tensor done, int64_t max_episode_steps,
                       int64_t seed, torch::Tensor draw) {
  int64_t B = state.size(0);
  float* sp = state.data_ptr<float>();
  const int64_t* ap = action.data_ptr<int64_t>();
  int32_t* scp = step_count.data_ptr<int32_t>();
  float* erp = ep_return.data_ptr<float>();
  int32_t* elp = ep_length.data_ptr<int32_t>();
  float* lerp = last_ep_return.data_ptr<float>();
  int32_t* lelp = last_ep_length.data_ptr<int32_t>();
  float* op = obs.data_ptr<float>();
  float* nop = next_obs.data_ptr<float>();
  float* rp = reward.data_ptr<float>();
  float* dp = discount.data_ptr<float>();
  uint8_t* stp = steptype.data_ptr<uint8_t>();
  uint8_t* dnp = done.data_ptr<uint8_t>();
  int32_t dr = *draw.data_ptr<int32_t>();

  // serial per-call stepping (envpool design): parallelism comes from the
  // ACTOR THREADS, each stepping its own pool with the GIL released — the
  // shared intra-op pool (at::parallel_for) collapses on big hosts (256
  // cores: 23.8K SPS vs 353K serial, and worker-thread OMP regions spawn
  // per-caller teams)
  auto serial_loop = [&](int64_t lo, int64_t hi) {
    for (int64_t b = lo; b < hi; ++b) {
      float* s = sp + b * SDIM;
      int a = (int)ap[b];
      if (a < 0) a = 0;
      if (a > 3) a = 3;
      float px = s[0] + PADDLE_SPEED * ((a == 2) - (a == 3));
      if (px < PADDLE_W / 2.0f) px = PADDLE_W / 2.0f;
      if (px > W - PADDLE_W / 2.0f) px = W - PADDLE_W / 2.0f;
      float bx = s[1] + s[3], by = s[2] + s[4];
      float vx = s[3], vy = s[4];
      if (bx < 1.0f || bx > W - 2.0f) vx = -vx;
      if (bx < 1.0f) bx = 1.0f;
      if (bx > W - 2.0f) bx = W - 2.0f;
      if (by < 1.0f) { vy = std::fabs(vy); by = 1.0f; }
      bool on_paddle = by >= PADDLE_Y - 1 && by <= PADDLE_Y + 1 &&
                       std::fabs(bx - px) <= PADDLE_W / 2.0f && vy > 0;
      if (on_paddle) {
        vx += 0.4f * (bx - px) / (PADDLE_W / 2.0f);
        vy = -std::fabs(vy);
      }
      float rew = 0.0f;
      if (by >= BRICK_TOP && by < BRICK_TOP + BRICK_ROWS * BRICK_H) {
        int br = (int)((by - BRICK_TOP) / BRICK_H);
        if (br < 0) br = 0;
        if (br > BRICK_ROWS - 1) br = BRICK_ROWS - 1;
        int bc = (int)(bx / BRICK_W);
        if (bc < 0) bc = 0;
        if (bc > BRICK_COLS - 1) bc = BRICK_COLS - 1;
        int k = br * BRICK_COLS + bc;
        if (s[SOFF + k] > 0.0f) {
          s[SOFF + k] = 0.0f;
          vy = -vy;
          rew = 1.0f;
        }
      }
      if (vx > 2.5f) vx = 2.5f;
      if (vx < -2.5f) vx = -2.5f;
      bool missed = by > H - 2.0f;
      float bricks_left = 0.0f;
      for (int k = 0; k < NBRICK; ++k) bricks_left += s[SOFF + k];
      bool terminated = missed || bricks_left <= 0.0f;
      s[0] = px; s[1] = bx;
      s[2] = by < 0.0f ? 0.0f : (by > H - 1.0f ? (float)(H - 1) : by);
      s[3] = vx; s[4] = vy;

      int sc = scp[b] + 1;
      bool truncated = (sc >= max_episode_steps) && !terminated;
      bool dn = terminated || truncated;
      float ret = erp[b] + rew;
      int len = elp[b] + 1;
      if (dn) { lerp[b] = ret; lelp[b] = len; }
      render(s, nop + b * H * W);  // true final obs (extras["next_obs"])
      if (dn) {
        reset_env(s, (uint64_t)seed, (uint64_t)b * 977 + 13, (uint64_t)dr);
        sc = 0; ret = 0.0f; len = 0;
        render(s, op + b * H * W);
      } else {
        std::memcpy(op + b * H * W, nop + b * H * W, sizeof(float) * H * W);
      }
      scp[b] = sc; erp[b] = ret; elp[b] = len;
      rp[b] = rew;
      dp[b] = terminated ? 0.0f : 1.0f;
      stp[b] = terminated ? ST_TERMINATED : (truncated ? ST_TRUNCATED : ST_MID);
      dnp[b] = dn ? 1 : 0;
    }
  };
  serial_loop(0, B);
  *draw.data_ptr<int32_t>() = dr + 1;
}

// ----------------------------------------------------------------- Pong
// Atari-class Pong: player paddle on the right, scripted opponent on the
// left (tracks the ball with capped speed), +1 when the opponent misses,
// -1 when the player misses; first to 21 points ends the episode.
namespace pong {

constexpr int PADDLE_H = 14;
constexpr float PLAYER_X = 80.0f, OPP_X = 3.0f;
constexpr float PADDLE_SPEED = 2.5f, OPP_SPEED = 1.6f, BSPEED = 1.6f;
// state row: player_y, opp_y, ball_x, ball_y, vx, vy, p_score, o_score
constexpr int SDIM = 8;

inline void reset_point(float* s, uint64_t seed, uint64_t env, uint64_t d) {
  s[2] = W / 2.0f;
  s[3] = 12.0f + hash_uniform(seed, env * 3, d) * 60.0f;
  s[4] = (hash_uniform(seed, env * 3 + 1, d) > 0.5f) ? BSPEED : -BSPEED;
  s[5] = (hash_uniform(seed, env * 3 + 2, d) - 0.5f) * 2.0f;
}

inline void reset_env(float* s, uint64_t seed, uint64_t env, uint64_t d) {
  s[0] = H / 2.0f;
  s[1] = H / 2.0f;
  s[6] = 0.0f;
  s[7] = 0.0f;
  reset_point(s, seed, env, d);
}

inline void render(const float* s, float* obs) {
  std::memset(obs, 0, sizeof(float) * H * W);
  for (int y = 0; y < H; y += 4) obs[y * W + W / 2] = 0.3f;  // net
  auto draw_paddle = [&](float py, int x) {
    int y0 = (int)py - PADDLE_H / 2;
    for (int y = y0; y < y0 + PADDLE_H; ++y) {
      if (y < 0 || y > H - 1) continue;
      obs[y * W + x] = 1.0f;
      obs[y * W + x + 1] = 1.0f;
    }
  };
  draw_paddle(s[0], (int)PLAYER_X);
  draw_paddle(s[1], (int)OPP_X);
  int by = (int)s[3], bx = (int)s[2];
  if (by < 0) by = 0;
  if (by > H - 2) by = H - 2;
  if (bx < 0) bx = 0;
  if (bx > W - 2) bx = W - 2;
  for (int dy = 0; dy < 2; ++dy)
    for (int dx = 0; dx < 2; ++dx) obs[(by + dy) * W + bx + dx] = 1.0f;
}

}  // namespace pong

void pong_cpu_reset(torch::Tensor state, torch::Tensor obs, int64_t seed,
                    int64_t draw) {
  TORCH_CHECK(state.size(1) == pong::SDIM, "state must be [B, 8]");
  int64_t B = state.size(0);
  float* sp = state.data_ptr<float>();
  float* op = obs.data_ptr<float>();
  // serial per-call stepping (envpool design): parallelism comes from the
  // ACTOR THREADS, each stepping its own pool with the GIL released — the
  // shared intra-op pool (at::parallel_for) collapses on big hosts (256
  // cores: 23.8K SPS vs 353K serial, and worker-thread OMP regions spawn
  // per-caller teams)
  auto serial_loop = [&](int64_t lo, int64_t hi) {
    for (int64_t b = lo; b < hi; ++b) {
      pong::reset_env(sp + b * pong::SDIM, (uint64_t)seed, (uint64_t)b,
                      (uint64_t)draw);
      pong::render(sp + b * pong::SDIM, op + b * H * W);
    }
  };
  serial_loop(0, B);
}

void pong_cpu_step(torch::Tensor state, torch::Tensor action,
                   torch::Tensor step_count, torch::Tensor ep_return,
                   torch::Tensor ep_length, torch::Tensor last_ep_return,
                   torch::Tensor last_ep_length, torch::Tensor obs,
                   torch::Tensor next_obs, torch::Tensor reward,
                   torch::Tensor discount, torch::Tensor steptype,
                   torch::Tensor done, int64_t max_episode_steps,
                   int64_t seed, torch::Tensor draw) {
  int64_t B = state.size(0);
  float* sp = state.data_ptr<float>();
  const int64_t* ap = action.data_ptr<int64_t>();
  int32_t* scp = step_count.data_ptr<int32_t>();
  float* erp = ep_return.data_ptr<float>();
  int32_t* elp = ep_length.data_ptr<int32_t>();
  float* lerp = last_ep_return.data_ptr<float>();
  int32_t* lelp = last_ep_length.data_ptr<int32_t>();
  float* op = obs.data_ptr<float>();
  float* nop = next_obs.data_ptr<float>();
  float* rp = reward.data_ptr<float>();
  float* dp = discount.data_ptr<float>();
  uint8_t* stp = steptype.data_ptr<uint8_t>();
  uint8_t* dnp = done.data_ptr<uint8_t>();
  int32_t dr = *draw.data_ptr<int32_t>();

  // serial per-call stepping (envpool design): parallelism comes from the
  // ACTOR THREADS, each stepping its own pool with the GIL released — the
  // shared intra-op pool (at::parallel_for) collapses on big hosts (256
  // cores: 23.8K SPS vs 353K serial, and worker-thread OMP regions spawn
  // per-caller teams)
  auto serial_loop = [&](int64_t lo, int64_t hi) {
    for (int64_t b = lo; b < hi; ++b) {
      float* s = sp + b * pong::SDIM;
      int a = (int)ap[b];  // 0 noop, 1 up, 2 down
      float py = s[0] + pong::PADDLE_SPEED * ((a == 2) - (a == 1));
      if (py < pong::PADDLE_H / 2.0f) py = pong::PADDLE_H / 2.0f;
      if (py > H - pong::PADDLE_H / 2.0f) py = H - pong::PADDLE_H / 2.0f;
      // opponent tracks the ball with capped speed
      float oy = s[1];
      float track = s[3] - oy;
      if (track > pong::OPP_SPEED) track = pong::OPP_SPEED;
      if (track < -pong::OPP_SPEED) track = -pong::OPP_SPEED;
      oy += track;
      float bx = s[2] + s[4], by = s[3] + s[5];
      float vx = s[4], vy = s[5];
      if (by < 1.0f) { vy = std::fabs(vy); by = 1.0f; }
      if (by > H - 2.0f) { vy = -std::fabs(vy); by = H - 2.0f; }
      // paddle bounces (with english)
      if (vx > 0 && bx >= pong::PLAYER_X - 1 && bx <= pong::PLAYER_X + 2 &&
          std::fabs(by - py) <= pong::PADDLE_H / 2.0f + 1) {
        vx = -std::fabs(vx) * 1.02f;
        vy += 0.25f * (by - py) / (pong::PADDLE_H / 2.0f);
        bx = pong::PLAYER_X - 1;
      }
      if (vx < 0 && bx <= pong::OPP_X + 2 && bx >= pong::OPP_X - 1 &&
          std::fabs(by - oy) <= pong::PADDLE_H / 2.0f + 1) {
        vx = std::fabs(vx);
        vy += 0.25f * (by - oy) / (pong::PADDLE_H / 2.0f);
        bx = pong::OPP_X + 2;
      }
      if (vx > 2.8f) vx = 2.8f;
      if (vx < -2.8f) vx = -2.8f;
      float rew = 0.0f;
      bool point = false;
      if (bx > W - 1.0f) { rew = -1.0f; s[7] += 1.0f; point = true; }
      if (bx < 0.0f) { rew = 1.0f; s[6] += 1.0f; point = true; }
      s[0] = py; s[1] = oy; s[2] = bx; s[3] = by; s[4] = vx; s[5] = vy;
      bool terminated = s[6] >= 21.0f || s[7] >= 21.0f;
      if (point && !terminated) pong::reset_point(s, (uint64_t)seed,
                                            (uint64_t)b * 131 + 7,
                                            (uint64_t)dr + (uint64_t)scp[b]);
      int sc = scp[b] + 1;
      bool truncated = (sc >= max_episode_steps) && !terminated;
      bool dn = terminated || truncated;
      float ret = erp[b] + rew;
      int len = elp[b] + 1;
      if (dn) { lerp[b] = ret; lelp[b] = len; }
      pong::render(s, nop + b * H * W);
      if (dn) {
        pong::reset_env(s, (uint64_t)seed, (uint64_t)b * 977 + 13, (uint64_t)dr);
        sc = 0; ret = 0.0f; len = 0;
        pong::render(s, op + b * H * W);
      } else {
        std::memcpy(op + b * H * W, nop + b * H * W, sizeof(float) * H * W);
      }
      scp[b] = sc; erp[b] = ret; elp[b] = len;
      rp[b] = rew;
      dp[b] = terminated ? 0.0f : 1.0f;
      stp[b] = terminated ? ST_TERMINATED : (truncated ? ST_TRUNCATED : ST_MID);
      dnp[b] = dn ? 1 : 0;
    }
  };
  serial_loop(0, B);
  *draw.data_ptr<int32_t>() = dr + 1;
}


// --------------------------------------------------- SpaceInvaders-class
// Alien grid marches and descends; the cannon fires one shot at a time;
// a random alive alien drops bombs. +1 per alien; clearing the wave pays
// +5 and terminates; a bomb hit or aliens reaching the cannon row ends
// the episode.
namespace spaceinv {
constexpr int AR = 5, AC = 8, NA = AR * AC;
// state: player_x, cooldown, shot_x, shot_y, shot_alive,
//        bomb_x, bomb_y, bomb_alive, adir, aoffx, aoffy, aliens[NA]
constexpr int SOFF = 11;
constexpr int SDIM = SOFF + NA;
constexpr float PSPEED = 2.5f, SHOT_V = 3.0f, BOMB_V = 1.2f;
constexpr int PW = 8, PY = 80;

inline void reset_env(float* s, uint64_t seed, uint64_t env, uint64_t draw) {
  s[0] = W / 2.0f;
  s[1] = 0.0f;
  s[2] = s[3] = 0.0f; s[4] = 0.0f;
  s[5] = s[6] = 0.0f; s[7] = 0.0f;
  s[8] = (hash_uniform(seed, env * 3, draw) > 0.5f) ? 1.0f : -1.0f;
  s[9] = 6.0f;
  s[10] = 8.0f;
  for (int k = 0; k < NA; ++k) s[SOFF + k] = 1.0f;
}

inline void render(const float* s, float* obs) {
  std::memset(obs, 0, sizeof(float) * H * W);
  // aliens: 5x3 blocks on a 9x7 lattice from (aoffx, aoffy)
  for (int r = 0; r < AR; ++r)
    for (int c = 0; c < AC; ++c) {
      if (s[SOFF + r * AC + c] <= 0.0f) continue;
      int y0 = (int)s[10] + r * 7, x0 = (int)s[9] + c * 9;
      for (int y = y0; y < y0 + 3; ++y)
        for (int x = x0; x < x0 + 5; ++x)
          if (y >= 0 && y < H && x >= 0 && x < W) obs[y * W + x] = 0.6f;
    }
  // player cannon
  int px = (int)s[0];
  for (int dx = -(PW / 2); dx < PW / 2; ++dx) {
    int x = px + dx;
    if (x < 0) x = 0;
    if (x > W - 1) x = W - 1;
    obs[PY * W + x] = 1.0f;
    obs[(PY + 1) * W + x] = 1.0f;
  }
  obs[(PY - 1) * W + (px < 0 ? 0 : (px > W - 1 ? W - 1 : px))] = 1.0f;
  if (s[4] > 0.0f) {  // shot, 1x2
    int x = (int)s[2], y = (int)s[3];
    if (x >= 0 && x < W && y >= 1 && y < H) {
      obs[y * W + x] = 0.9f;
      obs[(y - 1) * W + x] = 0.9f;
    }
  }
  if (s[7] > 0.0f) {  // bomb, 2x2
    int x = (int)s[5], y = (int)s[6];
    if (x >= 0 && x < W - 1 && y >= 0 && y < H - 1)
      for (int dy = 0; dy < 2; ++dy)
        for (int dx = 0; dx < 2; ++dx) obs[(y + dy) * W + x + dx] = 0.8f;
  }
}

void spaceinv_cpu_reset(torch::Tensor state, torch::Tensor obs,
                        int64_t seed, int64_t draw) {
  int64_t B = state.size(0);
  float* sp = state.data_ptr<float>();
  float* op = obs.data_ptr<float>();
  for (int64_t b = 0; b < B; ++b) {
    reset_env(sp + b * SDIM, (uint64_t)seed, (uint64_t)b, (uint64_t)draw);
    render(sp + b * SDIM, op + b * H * W);
  }
}

void spaceinv_cpu_step(torch::Tensor state, torch::Tensor action,
                       torch::Tensor step_count, torch::Tensor ep_return,
                       torch::Tensor ep_length, torch::Tensor last_ep_return,
                       torch::Tensor last_ep_length, torch::Tensor obs,
                       torch::Tensor next_obs, torch::Tensor reward,
                       torch::Tensor discount, torch::Tensor steptype,
                       torch::Tensor done, int64_t max_episode_steps,
                       int64_t seed, torch::Tensor draw) {
  int64_t B = state.size(0);
  float* sp = state.data_ptr<float>();
  const int64_t* ap = action.data_ptr<int64_t>();
  int32_t* scp = step_count.data_ptr<int32_t>();
  float* erp = ep_return.data_ptr<float>();
  int32_t* elp = ep_length.data_ptr<int32_t>();
  float* lerp = last_ep_return.data_ptr<float>();
  int32_t* lelp = last_ep_length.data_ptr<int32_t>();
  float* op = obs.data_ptr<float>();
  float* nop = next_obs.data_ptr<float>();
  float* rp = reward.data_ptr<float>();
  float* dp = discount.data_ptr<float>();
  uint8_t* stp = steptype.data_ptr<uint8_t>();
  uint8_t* dnp = done.data_ptr<uint8_t>();
  int32_t dr = *draw.data_ptr<int32_t>();
  for (int64_t b = 0; b < B; ++b) {
    float* s = sp + b * SDIM;
    int a = (int)ap[b];
    if (a < 0) a = 0;
    if (a > 3) a = 3;
    float px = s[0] + PSPEED * ((a == 2) - (a == 1));
    if (px < PW / 2.0f) px = PW / 2.0f;
    if (px > W - PW / 2.0f) px = W - PW / 2.0f;
    float cool = s[1] > 0.0f ? s[1] - 1.0f : 0.0f;
    float rew = 0.0f;
    // fire
    if (a == 3 && s[4] <= 0.0f && cool <= 0.0f) {
      s[2] = px; s[3] = PY - 2.0f; s[4] = 1.0f; cool = 6.0f;
    }
    // alien march
    float adir = s[8], aoffx = s[9], aoffy = s[10];
    aoffx += adir * 0.5f;
    float span = AC * 9.0f;
    if (aoffx < 2.0f || aoffx + span > W - 2.0f) { adir = -adir; aoffy += 3.0f; }
    // shot flight + alien hit
    if (s[4] > 0.0f) {
      s[3] -= SHOT_V;
      if (s[3] < 1.0f) s[4] = 0.0f;
      int c = (int)((s[2] - aoffx) / 9.0f);
      int r = (int)((s[3] - aoffy) / 7.0f);
      if (s[4] > 0.0f && r >= 0 && r < AR && c >= 0 && c < AC) {
        float rel_x = s[2] - (aoffx + c * 9.0f);
        float rel_y = s[3] - (aoffy + r * 7.0f);
        if (rel_x >= 0 && rel_x < 5.0f && rel_y >= 0 && rel_y < 3.0f &&
            s[SOFF + r * AC + c] > 0.0f) {
          s[SOFF + r * AC + c] = 0.0f;
          s[4] = 0.0f;
          rew += 1.0f;
        }
      }
    }
    // bombs
    bool player_hit = false;
    if (s[7] > 0.0f) {
      s[6] += BOMB_V;
      if (s[6] >= PY - 1.0f && std::fabs(s[5] - px) <= PW / 2.0f) {
        player_hit = true;
        s[7] = 0.0f;
      } else if (s[6] > H - 2.0f) {
        s[7] = 0.0f;
      }
    } else if (hash_uniform((uint64_t)seed, (uint64_t)b * 31 + 7,
                            (uint64_t)(dr + scp[b])) < 0.06f) {
      // drop from a random alive alien
      int pick = (int)(hash_uniform((uint64_t)seed, (uint64_t)b * 31 + 8,
                                    (uint64_t)(dr + scp[b])) * NA);
      for (int k = 0; k < NA; ++k) {
        int idx = (pick + k) % NA;
        if (s[SOFF + idx] > 0.0f) {
          int r = idx / AC, c = idx % AC;
          s[5] = aoffx + c * 9.0f + 2.0f;
          s[6] = aoffy + r * 7.0f + 3.0f;
          s[7] = 1.0f;
          break;
        }
      }
    }
    float alive = 0.0f;
    for (int k = 0; k < NA; ++k) alive += s[SOFF + k];
    bool landed = aoffy + AR * 7.0f >= PY - 2.0f;
    bool cleared = alive <= 0.0f;
    if (cleared) rew += 5.0f;
    bool terminated = player_hit || landed || cleared;
    s[0] = px; s[1] = cool; s[8] = adir; s[9] = aoffx; s[10] = aoffy;

    int sc = scp[b] + 1;
    bool truncated = (sc >= max_episode_steps) && !terminated;
    bool dn = terminated || truncated;
    float ret = erp[b] + rew;
    int len = elp[b] + 1;
    if (dn) { lerp[b] = ret; lelp[b] = len; }
    render(s, nop + b * H * W);
    if (dn) {
      reset_env(s, (uint64_t)seed, (uint64_t)b * 977 + 13, (uint64_t)dr);
      sc = 0; ret = 0.0f; len = 0;
      render(s, op + b * H * W);
    } else {
      std::memcpy(op + b * H * W, nop + b * H * W, sizeof(float) * H * W);
    }
    scp[b] = sc; erp[b] = ret; elp[b] = len;
    rp[b] = rew;
    dp[b] = terminated ? 0.0f : 1.0f;
    stp[b] = terminated ? ST_TERMINATED : (truncated ? ST_TRUNCATED : ST_MID);
    dnp[b] = dn ? 1 : 0;
  }
  *draw.data_ptr<int32_t>() = dr + 1;
}
}  // namespace spaceinv

// --------------------------------------------------------- Qbert-class
// Triangular pyramid of cubes; diagonal hops colour cubes (+1 first
// visit); colouring every cube pays +5 and terminates; hopping off the
// pyramid or colliding with the bouncing ball ends the episode.
namespace qbert {
constexpr int ROWS = 7, NCUBE = ROWS * (ROWS + 1) / 2;
// state: agent_r, agent_c, ball_r, ball_c, ball_alive, ball_timer,
//        colored[NCUBE]
constexpr int SOFF = 6;
constexpr int SDIM = SOFF + NCUBE;

inline int cube_index(int r, int c) { return r * (r + 1) / 2 + c; }

inline void reset_env(float* s, uint64_t seed, uint64_t env, uint64_t draw) {
  (void)seed; (void)env; (void)draw;
  s[0] = 0.0f; s[1] = 0.0f;
  s[2] = 0.0f; s[3] = 0.0f; s[4] = 0.0f;
  s[5] = 10.0f;
  for (int k = 0; k < NCUBE; ++k) s[SOFF + k] = 0.0f;
  s[SOFF + 0] = 1.0f;  // start cube coloured
}

inline void render(const float* s, float* obs) {
  std::memset(obs, 0, sizeof(float) * H * W);
  for (int r = 0; r < ROWS; ++r)
    for (int c = 0; c <= r; ++c) {
      int cx = W / 2 + (2 * c - r) * 6;
      int cy = 8 + r * 10;
      float v = s[SOFF + cube_index(r, c)] > 0.0f ? 0.8f : 0.35f;
      for (int y = cy; y < cy + 7; ++y)
        for (int x = cx - 4; x < cx + 4; ++x)
          if (y >= 0 && y < H && x >= 0 && x < W) obs[y * W + x] = v;
    }
  int ar = (int)s[0], ac = (int)s[1];
  int axc = W / 2 + (2 * ac - ar) * 6, ayc = 5 + ar * 10;
  for (int y = ayc; y < ayc + 4; ++y)
    for (int x = axc - 2; x < axc + 2; ++x)
      if (y >= 0 && y < H && x >= 0 && x < W) obs[y * W + x] = 1.0f;
  if (s[4] > 0.0f) {
    int br = (int)s[2], bc = (int)s[3];
    int bxc = W / 2 + (2 * bc - br) * 6, byc = 5 + br * 10;
    for (int y = byc; y < byc + 3; ++y)
      for (int x = bxc - 2; x < bxc + 1; ++x)
        if (y >= 0 && y < H && x >= 0 && x < W) obs[y * W + x] = 0.55f;
  }
}

void qbert_cpu_reset(torch::Tensor state, torch::Tensor obs, int64_t seed,
                     int64_t draw) {
  int64_t B = state.size(0);
  float* sp = state.data_ptr<float>();
  float* op = obs.data_ptr<float>();
  for (int64_t b = 0; b < B; ++b) {
    reset_env(sp + b * SDIM, (uint64_t)seed, (uint64_t)b, (uint64_t)draw);
    render(sp + b * SDIM, op + b * H * W);
  }
}

void qbert_cpu_step(torch::Tensor state, torch::Tensor action,
                    torch::Tensor step_count, torch::Tensor ep_return,
                    torch::Tensor ep_length, torch::Tensor last_ep_return,
                    torch::Tensor last_ep_length, torch::Tensor obs,
                    torch::Tensor next_obs, torch::Tensor reward,
                    torch::Tensor discount, torch::Tensor steptype,
                    torch::Tensor done, int64_t max_episode_steps,
                    int64_t seed, torch::Tensor draw) {
  int64_t B = state.size(0);
  float* sp = state.data_ptr<float>();
  const int64_t* ap = action.data_ptr<int64_t>();
  int32_t* scp = step_count.data_ptr<int32_t>();
  float* erp = ep_return.data_ptr<float>();
  int32_t* elp = ep_length.data_ptr<int32_t>();
  float* lerp = last_ep_return.data_ptr<float>();
  int32_t* lelp = last_ep_length.data_ptr<int32_t>();
  float* op = obs.data_ptr<float>();
  float* nop = next_obs.data_ptr<float>();
  float* rp = reward.data_ptr<float>();
  float* dp = discount.data_ptr<float>();
  uint8_t* stp = steptype.data_ptr<uint8_t>();
  uint8_t* dnp = done.data_ptr<uint8_t>();
  int32_t dr = *draw.data_ptr<int32_t>();
  for (int64_t b = 0; b < B; ++b) {
    float* s = sp + b * SDIM;
    int a = (int)ap[b];
    if (a < 0) a = 0;
    if (a > 3) a = 3;
    int r = (int)s[0], c = (int)s[1];
    // hops: 0 down-left, 1 down-right, 2 up-left, 3 up-right
    int nr = r + ((a < 2) ? 1 : -1);
    int nc = c + ((a == 1) ? 1 : 0) - ((a == 2) ? 1 : 0);
    float rew = 0.0f;
    bool fell = nr < 0 || nr >= ROWS || nc < 0 || nc > nr;
    bool cleared = false;
    if (!fell) {
      int k = cube_index(nr, nc);
      if (s[SOFF + k] <= 0.0f) {
        s[SOFF + k] = 1.0f;
        rew += 1.0f;
      }
      float colored = 0.0f;
      for (int q = 0; q < NCUBE; ++q) colored += s[SOFF + q];
      if (colored >= (float)NCUBE) { cleared = true; rew += 5.0f; }
      s[0] = (float)nr; s[1] = (float)nc;
    }
    // ball: spawns at the top after a countdown, hops down randomly
    bool caught = false;
    if (!fell && !cleared) {
      if (s[4] <= 0.0f) {
        s[5] -= 1.0f;
        if (s[5] <= 0.0f) { s[2] = 0.0f; s[3] = 0.0f; s[4] = 1.0f; }
      } else {
        int br = (int)s[2], bc = (int)s[3];
        int bnr = br + 1;
        int bnc = bc + ((hash_uniform((uint64_t)seed, (uint64_t)b * 53 + 5,
                                      (uint64_t)(dr + scp[b])) > 0.5f) ? 1 : 0);
        if (bnc > bnr) bnc = bnr;
        if (bnr >= ROWS) {
          s[4] = 0.0f;
          s[5] = 8.0f + 8.0f * hash_uniform((uint64_t)seed,
                                            (uint64_t)b * 53 + 6,
                                            (uint64_t)(dr + scp[b]));
        } else {
          s[2] = (float)bnr; s[3] = (float)bnc;
        }
      }
      caught = s[4] > 0.0f && (int)s[0] == (int)s[2] && (int)s[1] == (int)s[3];
    }
    bool terminated = fell || cleared || caught;

    int sc = scp[b] + 1;
    bool truncated = (sc >= max_episode_steps) && !terminated;
    bool dn = terminated || truncated;
    float ret = erp[b] + rew;
    int len = elp[b] + 1;
    if (dn) { lerp[b] = ret; lelp[b] = len; }
    render(s, nop + b * H * W);
    if (dn) {
      reset_env(s, (uint64_t)seed, (uint64_t)b * 977 + 13, (uint64_t)dr);
      sc = 0; ret = 0.0f; len = 0;
      render(s, op + b * H * W);
    } else {
      std::memcpy(op + b * H * W, nop + b * H * W, sizeof(float) * H * W);
    }
    scp[b] = sc; erp[b] = ret; elp[b] = len;
    rp[b] = rew;
    dp[b] = terminated ? 0.0f : 1.0f;
    stp[b] = terminated ? ST_TERMINATED : (truncated ? ST_TRUNCATED : ST_MID);
    dnp[b] = dn ? 1 : 0;
  }
  *draw.data_ptr<int32_t>() = dr + 1;
}
}  // namespace qbert


// --------------------------------------------------- VizDoom-basic-class
// First-person raycast shooter, the "basic" scenario: a rectangular room,
// a stationary monster at a random position along the far wall, the
// player strafes along the near wall and shoots. Rewards follow the
// vizdoom_basic shape: +101 kill, -5 per missed shot, -1 living penalty
// per step; episode ends on the kill or the step cap. The 84x84 obs is a
// true perspective render: per-column wall raycast (box walls, distance
// shading) with the monster as a distance-scaled billboard.
namespace vizdoom {
constexpr float ROOM_W = 8.0f, ROOM_D = 6.0f;
constexpr float PLAYER_Y = 0.5f, MONSTER_Y = 5.5f;
constexpr float FOV = 1.57079632679f;  // 90 degrees
constexpr float STRAFE = 0.35f, AIM_HALF = 0.06f;
constexpr float MONSTER_HALF_W = 0.35f;
// state: player_x, monster_x, monster_alive, cooldown
constexpr int SDIM = 4;

inline void reset_env(float* s, uint64_t seed, uint64_t env, uint64_t draw) {
  s[0] = ROOM_W / 2.0f;
  s[1] = 1.0f + hash_uniform(seed, env * 7 + 3, draw) * (ROOM_W - 2.0f);
  s[2] = 1.0f;
  s[3] = 0.0f;
}

inline void render(const float* s, float* obs) {
  const float px = s[0];
  const bool alive = s[2] > 0.0f;
  const float mx = s[1];
  // monster angular extent (from the player, facing +y)
  float mdx0 = (mx - MONSTER_HALF_W) - px, mdx1 = (mx + MONSTER_HALF_W) - px;
  float mdy = MONSTER_Y - PLAYER_Y;
  float mang0 = atan2f(mdx0, mdy), mang1 = atan2f(mdx1, mdy);
  float mdist = mdy;  // billboard depth (player faces +y)
  for (int c = 0; c < W; ++c) {
    float ang = ((float)c / (float)(W - 1) - 0.5f) * FOV;
    float dx = sinf(ang), dy = cosf(ang);
    // box-wall intersection from (px, PLAYER_Y) along (dx, dy), dy > 0
    float t_far = (ROOM_D - PLAYER_Y) / dy;
    float t_side = 1e9f;
    if (dx > 1e-6f) t_side = (ROOM_W - px) / dx;
    else if (dx < -1e-6f) t_side = -px / dx;
    float t = t_far < t_side ? t_far : t_side;
    float dist = t * dy;  // perpendicular distance (no fisheye)
    if (dist < 0.3f) dist = 0.3f;
    int wall_h = (int)(60.0f / dist);
    if (wall_h > H) wall_h = H;
    int w0 = H / 2 - wall_h / 2, w1 = H / 2 + wall_h / 2;
    float shade = 0.55f - 0.05f * dist;
    if (shade < 0.15f) shade = 0.15f;
    bool monster_col = alive && ang >= mang0 && ang <= mang1 && mdist < dist;
    int m_h = (int)(50.0f / mdist);
    int m0 = H / 2 - m_h / 2, m1 = H / 2 + m_h / 2;
    for (int r = 0; r < H; ++r) {
      float v;
      if (monster_col && r >= m0 && r < m1) v = 0.95f;
      else if (r >= w0 && r < w1) v = shade;
      else if (r >= w1) v = 0.25f;  // floor
      else v = 0.05f;               // ceiling
      obs[r * W + c] = v;
    }
    // crosshair
  }
  obs[(H / 2) * W + W / 2] = 1.0f;
  obs[(H / 2 + 1) * W + W / 2] = 1.0f;
}

void vizdoom_cpu_reset(torch::Tensor state, torch::Tensor obs, int64_t seed,
                       int64_t draw) {
  int64_t B = state.size(0);
  float* sp = state.data_ptr<float>();
  float* op = obs.data_ptr<float>();
  for (int64_t b = 0; b < B; ++b) {
    reset_env(sp + b * SDIM, (uint64_t)seed, (uint64_t)b, (uint64_t)draw);
    render(sp + b * SDIM, op + b * H * W);
  }
}

void vizdoom_cpu_step(torch::Tensor state, torch::Tensor action,
                      torch::Tensor step_count, torch::Tensor ep_return,
                      torch::Tensor ep_length, torch::Tensor last_ep_return,
                      torch::Tensor last_ep_length, torch::Tensor obs,
                      torch::Tensor next_obs, torch::Tensor reward,
                      torch::Tensor discount, torch::Tensor steptype,
                      torch::Tensor done, int64_t max_episode_steps,
                      int64_t seed, torch::Tensor draw) {
  int64_t B = state.size(0);
  float* sp = state.data_ptr<float>();
  const int64_t* ap = action.data_ptr<int64_t>();
  int32_t* scp = step_count.data_ptr<int32_t>();
  float* erp = ep_return.data_ptr<float>();
  int32_t* elp = ep_length.data_ptr<int32_t>();
  float* lerp = last_ep_return.data_ptr<float>();
  int32_t* lelp = last_ep_length.data_ptr<int32_t>();
  float* op = obs.data_ptr<float>();
  float* nop = next_obs.data_ptr<float>();
  float* rp = reward.data_ptr<float>();
  float* dp = discount.data_ptr<float>();
  uint8_t* stp = steptype.data_ptr<uint8_t>();
  uint8_t* dnp = done.data_ptr<uint8_t>();
  int32_t dr = *draw.data_ptr<int32_t>();
  for (int64_t b = 0; b < B; ++b) {
    float* s = sp + b * SDIM;
    int a = (int)ap[b];
    if (a < 0) a = 0;
    if (a > 3) a = 3;
    float px = s[0] + STRAFE * ((a == 2) - (a == 1));
    if (px < 0.7f) px = 0.7f;
    if (px > ROOM_W - 0.7f) px = ROOM_W - 0.7f;
    float cool = s[3] > 0.0f ? s[3] - 1.0f : 0.0f;
    float rew = -1.0f;  // living penalty
    bool killed = false;
    if (a == 3 && cool <= 0.0f) {
      cool = 3.0f;
      // hit if the monster centre is inside the aim cone (facing +y)
      float aim = atan2f(s[1] - px, MONSTER_Y - PLAYER_Y);
      if (s[2] > 0.0f && fabsf(aim) <= AIM_HALF) {
        killed = true;
        s[2] = 0.0f;
        rew += 101.0f;
      } else {
        rew -= 5.0f;  // missed shot
      }
    }
    s[0] = px;
    s[3] = cool;
    bool terminated = killed;

    int sc = scp[b] + 1;
    bool truncated = (sc >= max_episode_steps) && !terminated;
    bool dn = terminated || truncated;
    float ret = erp[b] + rew;
    int len = elp[b] + 1;
    if (dn) { lerp[b] = ret; lelp[b] = len; }
    render(s, nop + b * H * W);
    if (dn) {
      reset_env(s, (uint64_t)seed, (uint64_t)b * 977 + 13, (uint64_t)dr);
      sc = 0; ret = 0.0f; len = 0;
      render(s, op + b * H * W);
    } else {
      std::memcpy(op + b * H * W, nop + b * H * W, sizeof(float) * H * W);
    }
    scp[b] = sc; erp[b] = ret; elp[b] = len;
    rp[b] = rew;
    dp[b] = terminated ? 0.0f : 1.0f;
    stp[b] = terminated ? ST_TERMINATED : (truncated ? ST_TRUNCATED : ST_MID);
    dnp[b] = dn ? 1 : 0;
  }
  *draw.data_ptr<int32_t>() = dr + 1;
}
}  // namespace vizdoom

// second-wave games (phoenix / battlezone / doubledunk / namethisgame)
// live in envpool_games2.cpp and register through this hook
void register_games2(pybind11::module_& m);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  // gil_scoped_release: the whole fused step runs WITHOUT the GIL so
  // Sebulba's learner thread and sibling actor threads keep running
  // while a batch of envs steps (the reference's envpool releases the
  // GIL for exactly this reason)
  m.def("pong_reset", &pong_cpu_reset,
        "batched Pong reset + render (CPU, threaded)",
        py::call_guard<py::gil_scoped_release>());
  m.def("pong_step", &pong_cpu_step,
        "batched Pong fused step: physics + opponent AI + metrics + autoreset + render",
        py::call_guard<py::gil_scoped_release>());
  m.attr("PONG_STATE_DIM") = pong::SDIM;
  m.def("breakout_reset", &breakout_cpu_reset,
        "batched Breakout reset + render (CPU, threaded)",
        py::call_guard<py::gil_scoped_release>());
  m.def("breakout_step", &breakout_cpu_step,
        "batched Breakout fused step: physics + metrics + autoreset + render",
        py::call_guard<py::gil_scoped_release>());
  m.attr("STATE_DIM") = SDIM;
  m.def("spaceinv_reset", &spaceinv::spaceinv_cpu_reset,
        "batched SpaceInvaders-class reset + render",
        py::call_guard<py::gil_scoped_release>());
  m.def("spaceinv_step", &spaceinv::spaceinv_cpu_step,
        "batched SpaceInvaders-class fused step",
        py::call_guard<py::gil_scoped_release>());
  m.attr("SPACEINV_STATE_DIM") = spaceinv::SDIM;
  m.def("qbert_reset", &qbert::qbert_cpu_reset,
        "batched Qbert-class reset + render",
        py::call_guard<py::gil_scoped_release>());
  m.def("qbert_step", &qbert::qbert_cpu_step, "batched Qbert-class fused step",
        py::call_guard<py::gil_scoped_release>());
  m.attr("QBERT_STATE_DIM") = qbert::SDIM;
  m.def("vizdoom_reset", &vizdoom::vizdoom_cpu_reset,
        "batched VizDoom-basic-class reset + raycast render",
        py::call_guard<py::gil_scoped_release>());
  m.def("vizdoom_step", &vizdoom::vizdoom_cpu_step,
        "batched VizDoom-basic-class fused step",
        py::call_guard<py::gil_scoped_release>());
  m.attr("VIZDOOM_STATE_DIM") = vizdoom::SDIM;
  register_games2(m);
}
