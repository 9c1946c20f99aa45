// Native batched CPU pool — second wave of Atari-class games.
//
// Completes the reference's envpool scenario coverage
// (/root/reference/stoix/configs/env/envpool/: phoenix, battlezone,
// doubledunk, namethisgame) with original capability-class games on the
// same fused-step contract as envpool_cpu.cpp: one C++ call does physics,
// scripted opponents, termination/truncation, episode metrics, autoreset
// and 84x84 frame rendering, serially per call with the GIL released
// (actor THREADS are the parallelism — envpool's own design).
#include <torch/extension.h>

#include <cmath>
#include <cstdint>
#include <cstring>

namespace games2 {

constexpr int H = 84, W = 84;
constexpr uint8_t ST_MID = 1, ST_TERMINATED = 2, ST_TRUNCATED = 3;

inline float hash_uniform(uint64_t seed, uint64_t a, uint64_t b) {
  uint64_t x = seed + 0x9e3779b97f4a7c15ull * (a + 1) + 0xbf58476d1ce4e5b9ull * (b + 1);
  x ^= x >> 30; x *= 0xbf58476d1ce4e5b9ull;
  x ^= x >> 27; x *= 0x94d049bb133111ebull;
  x ^= x >> 31;
  return (float)(x >> 40) * (1.0f / 16777216.0f);
}

inline void fill_rect(float* obs, int y0, int y1, int x0, int x1, float v) {
  if (y0 < 0) y0 = 0;
  if (x0 < 0) x0 = 0;
  if (y1 > H) y1 = H;
  if (x1 > W) x1 = W;
  for (int y = y0; y < y1; ++y)
    for (int x = x0; x < x1; ++x) obs[y * W + x] = v;
}

// Shared step driver: per-game logic is a lambda returning `terminated`
// and accumulating reward; the metrics/truncation/autoreset/render
// epilogue is identical across games (same semantics as envpool_cpu.cpp).
template <class ResetF, class StepF, class RenderF>
void drive(torch::Tensor& state, torch::Tensor& action,
           torch::Tensor& step_count, torch::Tensor& ep_return,
           torch::Tensor& ep_length, torch::Tensor& last_ep_return,
           torch::Tensor& last_ep_length, torch::Tensor& obs,
           torch::Tensor& next_obs, torch::Tensor& reward,
           torch::Tensor& discount, torch::Tensor& steptype,
           torch::Tensor& done, int64_t max_episode_steps, int64_t seed,
           torch::Tensor& draw, int sdim, ResetF resetf, StepF stepf,
           RenderF renderf) {
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
    float* s = sp + b * sdim;
    float rew = 0.0f;
    bool terminated = stepf(s, (int)ap[b], (uint64_t)b, (uint64_t)(dr + scp[b]), rew);
    int sc = scp[b] + 1;
    bool truncated = (sc >= max_episode_steps) && !terminated;
    bool dn = terminated || truncated;
    float ret = erp[b] + rew;
    int len = elp[b] + 1;
    if (dn) { lerp[b] = ret; lelp[b] = len; }
    renderf(s, nop + b * H * W);
    if (dn) {
      resetf(s, (uint64_t)seed, (uint64_t)b * 977 + 13, (uint64_t)dr);
      sc = 0; ret = 0.0f; len = 0;
      renderf(s, op + b * H * W);
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

template <class ResetF, class RenderF>
void drive_reset(torch::Tensor& state, torch::Tensor& obs, int64_t seed,
                 int64_t draw, int sdim, ResetF resetf, RenderF renderf) {
  int64_t B = state.size(0);
  float* sp = state.data_ptr<float>();
  float* op = obs.data_ptr<float>();
  for (int64_t b = 0; b < B; ++b) {
    resetf(sp + b * sdim, (uint64_t)seed, (uint64_t)b, (uint64_t)draw);
    renderf(sp + b * sdim, op + b * H * W);
  }
}

// ------------------------------------------------------- Phoenix-class
// Formation of birds that periodically SWOOP at the player; one shot at
// a time; clearing the birds summons a multi-hit MOTHERSHIP. +1 per
// bird, +2 per mothership hit, +10 for destroying it (terminates with
// the win); a swooping bird reaching the player ends the episode.
namespace phoenix {
constexpr int NB = 8;  // 2 rows x 4 cols
// state: player_x, cool, shot_x, shot_y, shot_alive, form_phase,
//        phase (0 birds / 1 mothership), mother_x, mother_hp,
//        birds[NB x (alive, swoop, x, y)]
constexpr int SOFF = 9;
constexpr int SDIM = SOFF + NB * 4;
constexpr float PSPEED = 2.5f, SHOT_V = 3.0f;
constexpr int PW = 8, PY = 80;

inline float slot_x(int idx, float phase) {
  return 14.0f + (idx % 4) * 18.0f + 8.0f * sinf(phase);
}
inline float slot_y(int idx) { return 12.0f + (idx / 4) * 9.0f; }

inline void reset_env(float* s, uint64_t seed, uint64_t env, uint64_t draw) {
  s[0] = W / 2.0f; s[1] = 0.0f;
  s[2] = s[3] = s[4] = 0.0f;
  s[5] = hash_uniform(seed, env * 11 + 1, draw) * 6.283f;
  s[6] = 0.0f; s[7] = W / 2.0f; s[8] = 5.0f;
  for (int k = 0; k < NB; ++k) {
    float* bd = s + SOFF + 4 * k;
    bd[0] = 1.0f; bd[1] = 0.0f;
    bd[2] = slot_x(k, s[5]); bd[3] = slot_y(k);
  }
}

inline void render(const float* s, float* obs) {
  std::memset(obs, 0, sizeof(float) * H * W);
  if (s[6] < 0.5f) {
    for (int k = 0; k < NB; ++k) {
      const float* bd = s + SOFF + 4 * k;
      if (bd[0] <= 0.0f) continue;
      int x0 = (int)bd[2] - 2, y0 = (int)bd[3] - 1;
      fill_rect(obs, y0, y0 + 3, x0, x0 + 5, bd[1] > 0.0f ? 0.85f : 0.6f);
    }
  } else {
    int mx = (int)s[7];
    fill_rect(obs, 6, 12, mx - 9, mx + 9, 0.9f);  // mothership hull
    fill_rect(obs, 12, 14, mx - 3, mx + 3, 0.7f); // core
  }
  int px = (int)s[0];
  fill_rect(obs, PY, PY + 2, px - PW / 2, px + PW / 2, 1.0f);
  obs[(PY - 1) * W + (px < 0 ? 0 : (px > W - 1 ? W - 1 : px))] = 1.0f;
  if (s[4] > 0.0f) {
    int x = (int)s[2], y = (int)s[3];
    if (x >= 0 && x < W && y >= 1 && y < H) {
      obs[y * W + x] = 0.9f;
      obs[(y - 1) * W + x] = 0.9f;
    }
  }
}

inline bool step_env(float* s, int a, uint64_t env, uint64_t t, float& rew,
                     uint64_t seed) {
  if (a < 0) a = 0;
  if (a > 3) a = 3;
  float px = s[0] + PSPEED * ((a == 2) - (a == 1));
  if (px < PW / 2.0f) px = PW / 2.0f;
  if (px > W - PW / 2.0f) px = W - PW / 2.0f;
  float cool = s[1] > 0.0f ? s[1] - 1.0f : 0.0f;
  if (a == 3 && s[4] <= 0.0f && cool <= 0.0f) {
    s[2] = px; s[3] = PY - 2.0f; s[4] = 1.0f; cool = 6.0f;
  }
  s[5] += 0.08f;
  bool player_hit = false, cleared = false;
  if (s[6] < 0.5f) {
    // bird phase
    for (int k = 0; k < NB; ++k) {
      float* bd = s + SOFF + 4 * k;
      if (bd[0] <= 0.0f) continue;
      if (bd[1] <= 0.0f) {
        bd[2] = slot_x(k, s[5]);
        bd[3] = slot_y(k);
        if (hash_uniform(seed, env * 97 + 11 + k, t) < 0.004f) bd[1] = 1.0f;
      } else {
        bd[3] += 1.5f;
        bd[2] += 1.2f * ((px > bd[2]) - (px < bd[2]));
        if (bd[3] >= PY - 2.0f) {
          if (std::fabs(bd[2] - px) < 5.0f) player_hit = true;
          bd[1] = 0.0f;  // pull back to formation next step
        }
      }
    }
    // shot vs birds
    if (s[4] > 0.0f) {
      s[3] -= SHOT_V;
      if (s[3] < 1.0f) s[4] = 0.0f;
      for (int k = 0; s[4] > 0.0f && k < NB; ++k) {
        float* bd = s + SOFF + 4 * k;
        if (bd[0] > 0.0f && std::fabs(s[2] - bd[2]) < 3.0f &&
            std::fabs(s[3] - bd[3]) < 3.0f) {
          bd[0] = 0.0f; s[4] = 0.0f; rew += 1.0f;
        }
      }
    }
    float alive = 0.0f;
    for (int k = 0; k < NB; ++k) alive += s[SOFF + 4 * k];
    if (alive <= 0.0f) { s[6] = 1.0f; s[7] = W / 2.0f; s[8] = 5.0f; }
  } else {
    // mothership phase: drifts with the formation oscillator
    s[7] = W / 2.0f + 24.0f * sinf(0.5f * s[5]);
    if (s[4] > 0.0f) {
      s[3] -= SHOT_V;
      if (s[3] < 1.0f) s[4] = 0.0f;
      if (s[4] > 0.0f && s[3] <= 14.0f && std::fabs(s[2] - s[7]) < 9.0f) {
        s[4] = 0.0f;
        s[8] -= 1.0f;
        rew += 2.0f;
        if (s[8] <= 0.0f) { rew += 10.0f; cleared = true; }
      }
    }
  }
  s[0] = px; s[1] = cool;
  return player_hit || cleared;
}
}  // namespace phoenix

// ---------------------------------------------------- Battlezone-class
// First-person tank on an open plane: rotate / drive / fire. One enemy
// tank at a time closes in and fires on a timer when in range; +10 per
// kill (a fresh enemy spawns), getting shot ends the episode at -1.
// Obs: horizon split, distance-scaled enemy billboard at its bearing,
// radar strip along the top, crosshair.
namespace battlezone {
// state: px, py, pang, cool, ex, ey, e_alive, e_timer
constexpr int SDIM = 8;
constexpr float FOV = 1.57079632679f;
constexpr float ROT = 0.12f, FWD = 0.4f, AIM_HALF = 0.08f;
constexpr float KILL_RANGE = 18.0f, ENEMY_RANGE = 12.0f;

inline float wrap_pi(float a) {
  while (a > 3.14159265f) a -= 6.2831853f;
  while (a < -3.14159265f) a += 6.2831853f;
  return a;
}

inline void spawn_enemy(float* s, uint64_t seed, uint64_t env, uint64_t t) {
  float ang = hash_uniform(seed, env * 131 + 17, t) * 6.2831853f;
  float dist = 8.0f + 6.0f * hash_uniform(seed, env * 131 + 18, t);
  s[4] = s[0] + dist * sinf(ang);
  s[5] = s[1] + dist * cosf(ang);
  s[6] = 1.0f;
  s[7] = 90.0f;
}

inline void reset_env(float* s, uint64_t seed, uint64_t env, uint64_t draw) {
  s[0] = 0.0f; s[1] = 0.0f;
  s[2] = hash_uniform(seed, env * 7 + 1, draw) * 6.2831853f;
  s[3] = 0.0f;
  spawn_enemy(s, seed, env, draw + 1000);
}

inline void render(const float* s, float* obs) {
  for (int r = 0; r < H; ++r) {
    float v = r < H / 2 ? 0.05f : 0.3f;
    for (int c = 0; c < W; ++c) obs[r * W + c] = v;
  }
  float dx = s[4] - s[0], dy = s[5] - s[1];
  float dist = std::sqrt(dx * dx + dy * dy);
  float rel = wrap_pi(std::atan2(dx, dy) - s[2]);
  if (s[6] > 0.0f && std::fabs(rel) < FOV / 2.0f && dist > 0.5f) {
    int c = (int)((rel / FOV + 0.5f) * (W - 1));
    int half = (int)(20.0f / dist);
    if (half < 1) half = 1;
    fill_rect(obs, H / 2 - half, H / 2 + half, c - half, c + half, 0.9f);
    // turret
    fill_rect(obs, H / 2 - half - half / 2, H / 2 - half, c - half / 2,
              c + half / 2, 0.8f);
  }
  // radar strip: enemy bearing as a blip along the top rows
  fill_rect(obs, 1, 3, 0, W, 0.12f);
  if (s[6] > 0.0f) {
    int rx = (int)((rel / 3.14159265f + 1.0f) * 0.5f * (W - 1));
    fill_rect(obs, 1, 3, rx - 1, rx + 1, 1.0f);
  }
  obs[(H / 2) * W + W / 2] = 1.0f;
  obs[(H / 2 + 1) * W + W / 2] = 1.0f;
}

inline bool step_env(float* s, int a, uint64_t env, uint64_t t, float& rew,
                     uint64_t seed) {
  if (a < 0) a = 0;
  if (a > 4) a = 4;
  s[2] = wrap_pi(s[2] + ROT * ((a == 2) - (a == 1)));
  if (a == 3) {
    s[0] += FWD * sinf(s[2]);
    s[1] += FWD * cosf(s[2]);
    if (s[0] > 20.0f) s[0] = 20.0f;
    if (s[0] < -20.0f) s[0] = -20.0f;
    if (s[1] > 20.0f) s[1] = 20.0f;
    if (s[1] < -20.0f) s[1] = -20.0f;
  }
  float cool = s[3] > 0.0f ? s[3] - 1.0f : 0.0f;
  float dx = s[4] - s[0], dy = s[5] - s[1];
  float dist = std::sqrt(dx * dx + dy * dy);
  float rel = wrap_pi(std::atan2(dx, dy) - s[2]);
  if (a == 4 && cool <= 0.0f) {
    cool = 5.0f;
    if (s[6] > 0.0f && std::fabs(rel) < AIM_HALF && dist < KILL_RANGE) {
      rew += 10.0f;
      spawn_enemy(s, seed, env, t);
      dx = s[4] - s[0]; dy = s[5] - s[1];
      dist = std::sqrt(dx * dx + dy * dy);
    }
  }
  s[3] = cool;
  bool shot_down = false;
  if (s[6] > 0.0f) {
    // enemy closes in slowly
    if (dist > 0.5f) {
      s[4] -= 0.05f * dx / dist;
      s[5] -= 0.05f * dy / dist;
    }
    s[7] -= 1.0f;
    if (s[7] <= 0.0f) {
      if (dist < ENEMY_RANGE) {
        shot_down = true;
        rew -= 1.0f;
      } else {
        s[7] = 60.0f;
      }
    }
  }
  return shot_down;
}
}  // namespace battlezone

// ---------------------------------------------------- DoubleDunk-class
// Top-down half-court basketball vs a scripted defender. On OFFENSE,
// drive to the hoop and shoot (+2 inside the arc; a defender touch is a
// steal). After a steal the opponent drives for the hoop — touch them to
// steal back, or concede -2. Fixed horizon; return = net points.
namespace dunk {
// state: px, py, def_x, def_y, mode (0 offense / 1 defense), ox, oy, cool
constexpr int SDIM = 8;
constexpr float HOOP_X = (float)(W / 2), HOOP_Y = 8.0f;
constexpr float ARC = 18.0f, PSPEED = 2.5f, DSPEED = 1.7f, OSPEED = 1.8f;

inline void reset_positions(float* s, uint64_t seed, uint64_t env, uint64_t t) {
  s[0] = W / 2.0f + (hash_uniform(seed, env * 19 + 3, t) - 0.5f) * 30.0f;
  s[1] = 70.0f;
  s[2] = W / 2.0f; s[3] = 34.0f;
  s[4] = 0.0f;
  s[5] = s[6] = 0.0f;
  s[7] = 0.0f;
}

inline void reset_env(float* s, uint64_t seed, uint64_t env, uint64_t draw) {
  reset_positions(s, seed, env, draw);
}

inline void render(const float* s, float* obs) {
  for (int k = 0; k < H * W; ++k) obs[k] = 0.25f;  // court
  fill_rect(obs, (int)HOOP_Y - 2, (int)HOOP_Y + 2, (int)HOOP_X - 4,
            (int)HOOP_X + 4, 0.9f);  // hoop/backboard
  // three-point arc: semicircle of radius ARC under the hoop
  for (int k = 0; k < 32; ++k) {
    float a = 3.14159f * (float)k / 31.0f;  // 0..pi sweeps left to right
    int x = (int)(HOOP_X + ARC * cosf(a));
    int y = (int)(HOOP_Y + ARC * sinf(a));
    if (x >= 0 && x < W && y >= 0 && y < H) obs[y * W + x] = 0.4f;
  }
  bool defense = s[4] > 0.5f;
  fill_rect(obs, (int)s[1] - 1, (int)s[1] + 2, (int)s[0] - 1, (int)s[0] + 2, 1.0f);
  fill_rect(obs, (int)s[3] - 1, (int)s[3] + 2, (int)s[2] - 1, (int)s[2] + 2, 0.55f);
  if (defense) {
    fill_rect(obs, (int)s[6] - 1, (int)s[6] + 2, (int)s[5] - 1, (int)s[5] + 2, 0.75f);
    // ball on the opponent
    obs[((int)s[6] - 2 < 0 ? 0 : (int)s[6] - 2) * W + (int)s[5]] = 0.95f;
  } else {
    // ball on the player
    int by = (int)s[1] - 2; if (by < 0) by = 0;
    obs[by * W + (int)s[0]] = 0.95f;
  }
}

inline bool step_env(float* s, int a, uint64_t env, uint64_t t, float& rew,
                     uint64_t seed) {
  if (a < 0) a = 0;
  if (a > 4) a = 4;
  // actions: 0 up, 1 down, 2 left, 3 right, 4 shoot/steal
  float mx = PSPEED * ((a == 3) - (a == 2));
  float my = PSPEED * ((a == 1) - (a == 0));
  if (a >= 4) { mx = 0.0f; my = 0.0f; }
  s[0] += mx; s[1] += my;
  if (s[0] < 2.0f) s[0] = 2.0f;
  if (s[0] > W - 3.0f) s[0] = W - 3.0f;
  if (s[1] < 2.0f) s[1] = 2.0f;
  if (s[1] > H - 3.0f) s[1] = H - 3.0f;
  bool defense = s[4] > 0.5f;
  if (!defense) {
    // defender tracks the player, staying between player and hoop
    float tx = 0.5f * (s[0] + HOOP_X), ty = 0.5f * (s[1] + HOOP_Y);
    float ddx = tx - s[2], ddy = ty - s[3];
    float dn = std::sqrt(ddx * ddx + ddy * ddy);
    if (dn > 0.5f) { s[2] += DSPEED * ddx / dn; s[3] += DSPEED * ddy / dn; }
    float pdx = s[0] - s[2], pdy = s[1] - s[3];
    if (std::sqrt(pdx * pdx + pdy * pdy) < 4.0f) {
      // steal: opponent takes it from here
      s[4] = 1.0f; s[5] = s[2]; s[6] = s[3];
      return false;
    }
    if (a == 4) {
      float hd = std::sqrt((s[0] - HOOP_X) * (s[0] - HOOP_X) +
                           (s[1] - HOOP_Y) * (s[1] - HOOP_Y));
      if (hd < 6.0f) {
        rew += 2.0f;  // dunk
        reset_positions(s, seed, env, t);
      } else if (hd < ARC &&
                 hash_uniform(seed, env * 41 + 9, t) < 1.1f - hd / ARC) {
        rew += 2.0f;  // jump shot
        reset_positions(s, seed, env, t);
      } else {
        // miss: turnover
        s[4] = 1.0f; s[5] = s[2]; s[6] = s[3];
      }
    }
  } else {
    // opponent drives at the hoop; touch them to steal back
    float odx = HOOP_X - s[5], ody = HOOP_Y - s[6];
    float on = std::sqrt(odx * odx + ody * ody);
    if (on > 0.5f) { s[5] += OSPEED * odx / on; s[6] += OSPEED * ody / on; }
    float pdx = s[0] - s[5], pdy = s[1] - s[6];
    if (std::sqrt(pdx * pdx + pdy * pdy) < 4.0f) {
      s[4] = 0.0f;  // stole it back where they stand
    } else if (on < 8.0f) {
      rew -= 2.0f;  // conceded
      reset_positions(s, seed, env, t);
    }
  }
  return false;  // fixed horizon
}
}  // namespace dunk

// -------------------------------------------------- NameThisGame-class
// Undersea shooter: trim the octopus tentacles growing down toward the
// diver (+0.5 per trim) and shoot the patrolling shark (+5). Any
// tentacle reaching the sea floor ends the episode.
namespace ntg {
constexpr int NT = 6;
// state: px, cool, shot_x, shot_y, shot_alive, shark_x, shark_dir,
//        shark_alive, shark_timer, tent_len[NT]
constexpr int SOFF = 9;
constexpr int SDIM = SOFF + NT;
constexpr float PSPEED = 2.5f, SHOT_V = 3.0f;
constexpr int PW = 8, PY = 78;
constexpr float TENT_TOP = 14.0f, FLOOR_Y = 74.0f, SHARK_Y = 42.0f;

inline float tent_x(int k) { return 12.0f + k * 12.0f; }

inline void reset_env(float* s, uint64_t seed, uint64_t env, uint64_t draw) {
  s[0] = W / 2.0f; s[1] = 0.0f;
  s[2] = s[3] = s[4] = 0.0f;
  s[5] = 10.0f + hash_uniform(seed, env * 13 + 2, draw) * (W - 20.0f);
  s[6] = (hash_uniform(seed, env * 13 + 3, draw) > 0.5f) ? 1.0f : -1.0f;
  s[7] = 1.0f; s[8] = 0.0f;
  for (int k = 0; k < NT; ++k)
    s[SOFF + k] = 6.0f + 10.0f * hash_uniform(seed, env * 13 + 4 + k, draw);
}

inline void render(const float* s, float* obs) {
  std::memset(obs, 0, sizeof(float) * H * W);
  fill_rect(obs, 10, 12, 0, W, 0.5f);  // octopus body line
  for (int k = 0; k < NT; ++k) {
    int x = (int)tent_x(k);
    int y1 = (int)(TENT_TOP + s[SOFF + k]);
    fill_rect(obs, (int)TENT_TOP, y1, x - 1, x + 1, 0.7f);
  }
  if (s[7] > 0.0f) {
    int sx = (int)s[5];
    fill_rect(obs, (int)SHARK_Y - 2, (int)SHARK_Y + 2, sx - 4, sx + 4, 0.9f);
    // tail
    fill_rect(obs, (int)SHARK_Y - 1, (int)SHARK_Y + 1,
              s[6] > 0 ? sx - 6 : sx + 4, s[6] > 0 ? sx - 4 : sx + 6, 0.9f);
  }
  int px = (int)s[0];
  fill_rect(obs, PY, PY + 3, px - PW / 2, px + PW / 2, 1.0f);
  fill_rect(obs, (int)FLOOR_Y + 8, H, 0, W, 0.2f);  // sea floor
  if (s[4] > 0.0f) {
    int x = (int)s[2], y = (int)s[3];
    if (x >= 0 && x < W && y >= 1 && y < H) {
      obs[y * W + x] = 0.95f;
      obs[(y - 1) * W + x] = 0.95f;
    }
  }
}

inline bool step_env(float* s, int a, uint64_t env, uint64_t t, float& rew,
                     uint64_t seed) {
  if (a < 0) a = 0;
  if (a > 3) a = 3;
  float px = s[0] + PSPEED * ((a == 2) - (a == 1));
  if (px < PW / 2.0f) px = PW / 2.0f;
  if (px > W - PW / 2.0f) px = W - PW / 2.0f;
  float cool = s[1] > 0.0f ? s[1] - 1.0f : 0.0f;
  if (a == 3 && s[4] <= 0.0f && cool <= 0.0f) {
    s[2] = px; s[3] = PY - 2.0f; s[4] = 1.0f; cool = 5.0f;
  }
  // tentacles grow at slightly different speeds
  for (int k = 0; k < NT; ++k)
    s[SOFF + k] += 0.05f + 0.05f * hash_uniform(seed, env * 61 + 21 + k, t);
  // shark patrol
  if (s[7] > 0.0f) {
    s[5] += 1.2f * s[6];
    if (s[5] < 6.0f || s[5] > W - 6.0f) s[6] = -s[6];
  } else {
    s[8] -= 1.0f;
    if (s[8] <= 0.0f) {
      s[7] = 1.0f;
      s[5] = s[6] > 0 ? 6.0f : W - 6.0f;
    }
  }
  // shot flight: shark first, then tentacle tips
  if (s[4] > 0.0f) {
    s[3] -= SHOT_V;
    if (s[3] < 1.0f) s[4] = 0.0f;
    if (s[4] > 0.0f && s[7] > 0.0f && std::fabs(s[3] - SHARK_Y) < 3.0f &&
        std::fabs(s[2] - s[5]) < 5.0f) {
      s[7] = 0.0f; s[8] = 40.0f; s[4] = 0.0f;
      rew += 5.0f;
    }
    for (int k = 0; s[4] > 0.0f && k < NT; ++k) {
      float tip = TENT_TOP + s[SOFF + k];
      if (std::fabs(s[2] - tent_x(k)) < 2.0f && s[3] <= tip &&
          s[3] > TENT_TOP) {
        s[SOFF + k] -= 8.0f;
        if (s[SOFF + k] < 2.0f) s[SOFF + k] = 2.0f;
        s[4] = 0.0f;
        rew += 0.5f;
      }
    }
  }
  s[0] = px; s[1] = cool;
  for (int k = 0; k < NT; ++k)
    if (TENT_TOP + s[SOFF + k] >= FLOOR_Y + 4.0f) return true;
  return false;
}
}  // namespace ntg

}  // namespace games2

#define POOL_BIND(NS, NAME)                                                    \
  void NAME##_cpu_reset(torch::Tensor state, torch::Tensor obs, int64_t seed,  \
                        int64_t draw) {                                        \
    games2::drive_reset(state, obs, seed, draw, games2::NS::SDIM,              \
                        games2::NS::reset_env, games2::NS::render);            \
  }                                                                            \
  void NAME##_cpu_step(                                                        \
      torch::Tensor state, torch::Tensor action, torch::Tensor step_count,     \
      torch::Tensor ep_return, torch::Tensor ep_length,                        \
      torch::Tensor last_ep_return, torch::Tensor last_ep_length,              \
      torch::Tensor obs, torch::Tensor next_obs, torch::Tensor reward,         \
      torch::Tensor discount, torch::Tensor steptype, torch::Tensor done,      \
      int64_t max_episode_steps, int64_t seed, torch::Tensor draw) {           \
    auto stepf = [seed](float* s, int a, uint64_t env, uint64_t t,             \
                        float& rew) {                                          \
      return games2::NS::step_env(s, a, env, t, rew, (uint64_t)seed);          \
    };                                                                         \
    games2::drive(state, action, step_count, ep_return, ep_length,             \
                  last_ep_return, last_ep_length, obs, next_obs, reward,       \
                  discount, steptype, done, max_episode_steps, seed, draw,     \
                  games2::NS::SDIM, games2::NS::reset_env, stepf,              \
                  games2::NS::render);                                         \
  }

POOL_BIND(phoenix, phoenix)
POOL_BIND(battlezone, battlezone)
POOL_BIND(dunk, doubledunk)
POOL_BIND(ntg, namethisgame)

void register_games2(pybind11::module_& m) {
  namespace py = pybind11;
  m.def("phoenix_reset", &phoenix_cpu_reset,
        "batched Phoenix-class reset + render",
        py::call_guard<py::gil_scoped_release>());
  m.def("phoenix_step", &phoenix_cpu_step, "batched Phoenix-class fused step",
        py::call_guard<py::gil_scoped_release>());
  m.attr("PHOENIX_STATE_DIM") = games2::phoenix::SDIM;
  m.def("battlezone_reset", &battlezone_cpu_reset,
        "batched Battlezone-class reset + render",
        py::call_guard<py::gil_scoped_release>());
  m.def("battlezone_step", &battlezone_cpu_step,
        "batched Battlezone-class fused step",
        py::call_guard<py::gil_scoped_release>());
  m.attr("BATTLEZONE_STATE_DIM") = games2::battlezone::SDIM;
  m.def("doubledunk_reset", &doubledunk_cpu_reset,
        "batched DoubleDunk-class reset + render",
        py::call_guard<py::gil_scoped_release>());
  m.def("doubledunk_step", &doubledunk_cpu_step,
        "batched DoubleDunk-class fused step",
        py::call_guard<py::gil_scoped_release>());
  m.attr("DOUBLEDUNK_STATE_DIM") = games2::dunk::SDIM;
  m.def("namethisgame_reset", &namethisgame_cpu_reset,
        "batched NameThisGame-class reset + render",
        py::call_guard<py::gil_scoped_release>());
  m.def("namethisgame_step", &namethisgame_cpu_step,
        "batched NameThisGame-class fused step",
        py::call_guard<py::gil_scoped_release>());
  m.attr("NAMETHISGAME_STATE_DIM") = games2::ntg::SDIM;
}
