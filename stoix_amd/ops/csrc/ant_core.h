// Ant physics as device functions, shared between the standalone env step
// kernel (envs.hip) and the fused rollout megakernel (mlp.hip): one env's
// whole control step (4 substeps), reset-state generation, and the
// state->observation packing. Keep EXACTLY in sync with
// stoix_amd/envs/ant.py (the torch reference used by the numerics tests).
#pragma once
#include "common.h"

#define ANT_STATE 29
#define ANT_OBS 27
#define ANT_ACT 8

struct V3 {
  float x, y, z;
};
DEV_INLINE V3 v3(float x, float y, float z) { return {x, y, z}; }
DEV_INLINE V3 add3(V3 a, V3 b) { return {a.x + b.x, a.y + b.y, a.z + b.z}; }
DEV_INLINE V3 scale3(V3 a, float s) { return {a.x * s, a.y * s, a.z * s}; }
DEV_INLINE V3 cross3(V3 a, V3 b) {
  return {a.y * b.z - a.z * b.y, a.z * b.x - a.x * b.z, a.x * b.y - a.y * b.x};
}
// rotate v by quaternion q = (w, x, y, z)
DEV_INLINE V3 quat_rot(const float* q, V3 v) {
  V3 qv = {q[1], q[2], q[3]};
  V3 uv = cross3(qv, v);
  V3 uuv = cross3(qv, uv);
  return add3(v, add3(scale3(uv, 2.0f * q[0]), scale3(uuv, 2.0f)));
}

// One full Ant control step (DT with SUBSTEPS) on a register-resident
// state; action must already be clamped to [-1, 1]. Writes reward and
// terminated; scrubs non-finite states.
DEV_INLINE void ant_physics_step(float* __restrict__ s,
                                 const float* __restrict__ a, float* reward,
                                 bool* terminated) {
  const float TORSO_MASS = 10.0f, TORSO_INERTIA = 0.4f;
  const float HIP_RADIUS = 0.2f, L1 = 0.2f, L2 = 0.4f;
  const float JOINT_INERTIA = 0.08f, JOINT_DAMPING = 1.2f, GEAR = 15.0f;
  const float HIP_LIMIT = 0.6f, KNEE_LO = 0.4f, KNEE_HI = 1.4f, LIMIT_K = 40.0f;
  const float KN = 2.0e3f, KD = 40.0f, FRICTION = 1.0f, GRAV = -9.81f;
  const float DT = 0.05f;
  const int SUBSTEPS = 4;
  const float CTRL_COST = 0.5f, CONTACT_COST = 5e-4f, HEALTHY = 1.0f;
  const float Z_MIN = 0.2f, Z_MAX = 1.0f;
  const float SQ2 = 0.70710678118654752f;
  const float hdx[4] = {SQ2, -SQ2, -SQ2, SQ2};
  const float hdy[4] = {SQ2, SQ2, -SQ2, -SQ2};
  // atan2f(hdy, hdx) is constant per leg: pi/4 + leg*pi/2
  const float base_angs[4] = {0.78539816339744831f, 2.35619449019234493f,
                              -2.35619449019234493f, -0.78539816339744831f};

  float x_before = s[0];
  float contact_mag = 0.0f;
  const float dt = DT / SUBSTEPS;

  for (int sub = 0; sub < SUBSTEPS; ++sub) {
    float* pos = s + 0;
    float* quat = s + 3;
    float* linvel = s + 7;
    float* angvel = s + 10;
    float* qpos = s + 13;
    float* qvel = s + 21;

    // foot positions & moment arms from the PRE-update joint state
    V3 r_arm[4], foot_w[4];
#pragma unroll
    for (int leg = 0; leg < 4; ++leg) {
      float leg_ang = base_angs[leg] + qpos[leg];
      float ca, sa, ck, sk;
      __sincosf(leg_ang, &sa, &ca);
      __sincosf(qpos[4 + leg], &sk, &ck);
      V3 body_off = {hdx[leg] * HIP_RADIUS + ca * L1 + ca * L2 * ck,
                     hdy[leg] * HIP_RADIUS + sa * L1 + sa * L2 * ck,
                     -L2 * sk};
      V3 r = quat_rot(quat, body_off);
      r_arm[leg] = r;
      foot_w[leg] = {pos[0] + r.x, pos[1] + r.y, pos[2] + r.z};
    }

    // joints: damped inertial with soft limits
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float q = qpos[j];
      float limit_tau;
      if (j < 4) {
        limit_tau =
            -LIMIT_K * (fmaxf(q - HIP_LIMIT, 0.0f) - fmaxf(-HIP_LIMIT - q, 0.0f));
      } else {
        limit_tau =
            -LIMIT_K * (fmaxf(q - KNEE_HI, 0.0f) - fmaxf(KNEE_LO - q, 0.0f));
      }
      float qacc = (GEAR * a[j] - JOINT_DAMPING * qvel[j] + limit_tau) / JOINT_INERTIA;
      qvel[j] += dt * qacc;
      qpos[j] += dt * qvel[j];
    }

    // contacts
    V3 total_f = {0, 0, 0}, total_tau = {0, 0, 0};
#pragma unroll
    for (int leg = 0; leg < 4; ++leg) {
      V3 r = r_arm[leg];
      V3 av = {angvel[0], angvel[1], angvel[2]};
      V3 fv = add3(v3(linvel[0], linvel[1], linvel[2]), cross3(av, r));
      float pen = fmaxf(-foot_w[leg].z, 0.0f);
      float fn = 0.0f;
      if (pen > 0.0f) fn = fmaxf(KN * pen - KD * fv.z, 0.0f);
      float ftx = -FRICTION * fn * tanhf(4.0f * fv.x);
      float fty = -FRICTION * fn * tanhf(4.0f * fv.y);
      V3 cf = {ftx, fty, fn};
      total_f = add3(total_f, cf);
      total_tau = add3(total_tau, cross3(r, cf));
      contact_mag += fabsf(cf.x) + fabsf(cf.y) + fabsf(cf.z);
    }

    // torso integration
    linvel[0] += dt * (total_f.x / TORSO_MASS);
    linvel[1] += dt * (total_f.y / TORSO_MASS);
    linvel[2] += dt * (total_f.z / TORSO_MASS + GRAV);
    pos[0] += dt * linvel[0];
    pos[1] += dt * linvel[1];
    pos[2] += dt * linvel[2];
    angvel[0] += dt * (total_tau.x / TORSO_INERTIA - 0.2f * angvel[0]);
    angvel[1] += dt * (total_tau.y / TORSO_INERTIA - 0.2f * angvel[1]);
    angvel[2] += dt * (total_tau.z / TORSO_INERTIA - 0.2f * angvel[2]);
    {
      float w = quat[0], qx = quat[1], qy = quat[2], qz = quat[3];
      float ox = angvel[0], oy = angvel[1], oz = angvel[2];
      float dw = 0.5f * (-qx * ox - qy * oy - qz * oz);
      float dx = 0.5f * (w * ox + qy * oz - qz * oy);
      float dy = 0.5f * (w * oy + qz * ox - qx * oz);
      float dz = 0.5f * (w * oz + qx * oy - qy * ox);
      w += dt * dw; qx += dt * dx; qy += dt * dy; qz += dt * dz;
      float n = sqrtf(fmaxf(w * w + qx * qx + qy * qy + qz * qz, 1e-16f));
      quat[0] = w / n; quat[1] = qx / n; quat[2] = qy / n; quat[3] = qz / n;
    }
    float torso_pen = fmaxf(0.12f - pos[2], 0.0f);
    linvel[2] += dt * KN / TORSO_MASS * torso_pen;
  }

  // reward & termination
  float forward_vel = (s[0] - x_before) / DT;
  float ctrl_cost = 0.0f;
#pragma unroll
  for (int i = 0; i < ANT_ACT; ++i) ctrl_cost += a[i] * a[i];
  ctrl_cost *= CTRL_COST;
  float cm = contact_mag / SUBSTEPS;
  float contact_cost = CONTACT_COST * cm * cm;
  bool finite = true;
#pragma unroll
  for (int i = 0; i < ANT_STATE; ++i) finite = finite && isfinite(s[i]);
  float z = s[2];
  bool healthy = (z > Z_MIN) && (z < Z_MAX) && finite;
  float r = forward_vel + HEALTHY - ctrl_cost - contact_cost;
  if (!isfinite(r)) r = 0.0f;
  *reward = r;
  *terminated = !healthy;
  if (!finite) {
#pragma unroll
    for (int i = 0; i < ANT_STATE; ++i) s[i] = isfinite(s[i]) ? s[i] : 0.0f;
  }
}

// Fresh episode state with philox reset noise (mirrors Ant._reset_fn; the
// same draws as the standalone kernel's autoreset: stream 1).
DEV_INLINE void ant_reset_state(float* __restrict__ s, uint64_t seed,
                                uint32_t b, uint32_t draw) {
  const float TORSO_Z0 = 0.55f;
  Rng4 r0 = philox_uniform4(seed, 1u, b, draw * 8u + 0u);
  Rng4 r1 = philox_uniform4(seed, 1u, b, draw * 8u + 1u);
  Rng4 r2 = philox_uniform4(seed, 1u, b, draw * 8u + 2u);
  Rng4 r3 = philox_uniform4(seed, 1u, b, draw * 8u + 3u);
  Rng4 r4 = philox_uniform4(seed, 1u, b, draw * 8u + 4u);
  float u[20] = {r0.a, r0.b, r0.c, r0.d, r1.a, r1.b, r1.c, r1.d,
                 r2.a, r2.b, r2.c, r2.d, r3.a, r3.b, r3.c, r3.d,
                 r4.a, r4.b, r4.c, r4.d};
#pragma unroll
  for (int i = 0; i < ANT_STATE; ++i) s[i] = 0.0f;
  s[2] = TORSO_Z0;
  s[3] = 1.0f;
#pragma unroll
  for (int i = 0; i < 8; ++i) s[13 + i] = -0.1f + 0.2f * u[i];
#pragma unroll
  for (int i = 0; i < 4; ++i) s[17 + i] += 0.9f;
#pragma unroll
  for (int i = 0; i < 8; ++i) s[21 + i] = -0.05f + 0.1f * u[8 + i];
#pragma unroll
  for (int i = 0; i < 3; ++i) s[7 + i] = -0.05f + 0.1f * u[16 + i];
}

DEV_INLINE void ant_write_obs(const float* __restrict__ s,
                              float* __restrict__ o) {
  o[0] = s[2];
  o[1] = s[3]; o[2] = s[4]; o[3] = s[5]; o[4] = s[6];
#pragma unroll
  for (int i = 0; i < 8; ++i) o[5 + i] = s[13 + i];
  o[13] = s[7]; o[14] = s[8]; o[15] = s[9];
  o[16] = s[10]; o[17] = s[11]; o[18] = s[12];
#pragma unroll
  for (int i = 0; i < 8; ++i) o[19 + i] = s[21 + i];
}

// Lane-parallel variant: FOUR lanes cooperate on one env (lane quad; leg =
// lane&3). Each lane holds a replicated copy of the 29-float state, does
// its own leg's foot/contact and its own hip+knee joint updates, and the
// quad exchanges joint updates / reduces contact forces with width-4
// shuffles. The torso integration is redundantly computed by all 4 lanes
// (identical results). ~2.5x lower per-env latency than the serial
// ant_physics_step; numerics differ only in the summation order of the 4
// legs' contact forces (butterfly vs serial, ~1 ulp).
DEV_INLINE void ant_physics_step_x4(float* __restrict__ s,
                                    const float* __restrict__ a, int leg,
                                    float* reward, bool* terminated) {
  const float TORSO_MASS = 10.0f, TORSO_INERTIA = 0.4f;
  const float HIP_RADIUS = 0.2f, L1 = 0.2f, L2 = 0.4f;
  const float JOINT_INERTIA = 0.08f, JOINT_DAMPING = 1.2f, GEAR = 15.0f;
  const float HIP_LIMIT = 0.6f, KNEE_LO = 0.4f, KNEE_HI = 1.4f, LIMIT_K = 40.0f;
  const float KN = 2.0e3f, KD = 40.0f, FRICTION = 1.0f, GRAV = -9.81f;
  const float DT = 0.05f;
  const int SUBSTEPS = 4;
  const float CTRL_COST = 0.5f, CONTACT_COST = 5e-4f, HEALTHY = 1.0f;
  const float Z_MIN = 0.2f, Z_MAX = 1.0f;
  const float SQ2 = 0.70710678118654752f;
  const float hdx[4] = {SQ2, -SQ2, -SQ2, SQ2};
  const float hdy[4] = {SQ2, SQ2, -SQ2, -SQ2};
  const float base_angs[4] = {0.78539816339744831f, 2.35619449019234493f,
                              -2.35619449019234493f, -0.78539816339744831f};

  float x_before = s[0];
  float contact_mag = 0.0f;
  const float dt = DT / SUBSTEPS;

  for (int sub = 0; sub < SUBSTEPS; ++sub) {
    float* pos = s + 0;
    float* quat = s + 3;
    float* linvel = s + 7;
    float* angvel = s + 10;
    float* qpos = s + 13;
    float* qvel = s + 21;

    // this lane's leg: foot position & moment arm (pre-update joints)
    float leg_ang = base_angs[leg] + qpos[leg];
    float ca, sa, ck, sk;
    __sincosf(leg_ang, &sa, &ca);
    __sincosf(qpos[4 + leg], &sk, &ck);
    V3 body_off = {hdx[leg] * HIP_RADIUS + ca * L1 + ca * L2 * ck,
                   hdy[leg] * HIP_RADIUS + sa * L1 + sa * L2 * ck,
                   -L2 * sk};
    V3 r = quat_rot(quat, body_off);
    V3 foot_w = {pos[0] + r.x, pos[1] + r.y, pos[2] + r.z};

    // this lane's joints (hip `leg`, knee `4+leg`); then exchange within
    // the quad so every lane has all 8 updated joints
    float hip_q, hip_v, knee_q, knee_v;
    {
      float q = qpos[leg];
      float lt = -LIMIT_K * (fmaxf(q - HIP_LIMIT, 0.0f) - fmaxf(-HIP_LIMIT - q, 0.0f));
      float qacc = (GEAR * a[leg] - JOINT_DAMPING * qvel[leg] + lt) / JOINT_INERTIA;
      hip_v = qvel[leg] + dt * qacc;
      hip_q = qpos[leg] + dt * hip_v;
      q = qpos[4 + leg];
      lt = -LIMIT_K * (fmaxf(q - KNEE_HI, 0.0f) - fmaxf(KNEE_LO - q, 0.0f));
      qacc = (GEAR * a[4 + leg] - JOINT_DAMPING * qvel[4 + leg] + lt) / JOINT_INERTIA;
      knee_v = qvel[4 + leg] + dt * qacc;
      knee_q = qpos[4 + leg] + dt * knee_v;
    }
#pragma unroll
    for (int jj = 0; jj < 4; ++jj) {
      qpos[jj] = __shfl(hip_q, jj, 4);
      qvel[jj] = __shfl(hip_v, jj, 4);
      qpos[4 + jj] = __shfl(knee_q, jj, 4);
      qvel[4 + jj] = __shfl(knee_v, jj, 4);
    }

    // this lane's leg contact; quad-reduce the force/torque sums
    V3 av = {angvel[0], angvel[1], angvel[2]};
    V3 fv = add3(v3(linvel[0], linvel[1], linvel[2]), cross3(av, r));
    float pen = fmaxf(-foot_w.z, 0.0f);
    float fn = 0.0f;
    if (pen > 0.0f) fn = fmaxf(KN * pen - KD * fv.z, 0.0f);
    V3 cf = {-FRICTION * fn * tanhf(4.0f * fv.x),
             -FRICTION * fn * tanhf(4.0f * fv.y), fn};
    V3 tq = cross3(r, cf);
    float cm = fabsf(cf.x) + fabsf(cf.y) + fabsf(cf.z);
#pragma unroll
    for (int m = 1; m < 4; m <<= 1) {
      cf.x += __shfl_xor(cf.x, m, 4);
      cf.y += __shfl_xor(cf.y, m, 4);
      cf.z += __shfl_xor(cf.z, m, 4);
      tq.x += __shfl_xor(tq.x, m, 4);
      tq.y += __shfl_xor(tq.y, m, 4);
      tq.z += __shfl_xor(tq.z, m, 4);
      cm += __shfl_xor(cm, m, 4);
    }
    contact_mag += cm;

    // torso integration (redundant across the quad; identical values)
    linvel[0] += dt * (cf.x / TORSO_MASS);
    linvel[1] += dt * (cf.y / TORSO_MASS);
    linvel[2] += dt * (cf.z / TORSO_MASS + GRAV);
    pos[0] += dt * linvel[0];
    pos[1] += dt * linvel[1];
    pos[2] += dt * linvel[2];
    angvel[0] += dt * (tq.x / TORSO_INERTIA - 0.2f * angvel[0]);
    angvel[1] += dt * (tq.y / TORSO_INERTIA - 0.2f * angvel[1]);
    angvel[2] += dt * (tq.z / TORSO_INERTIA - 0.2f * angvel[2]);
    {
      float w = quat[0], qx = quat[1], qy = quat[2], qz = quat[3];
      float ox = angvel[0], oy = angvel[1], oz = angvel[2];
      float dw = 0.5f * (-qx * ox - qy * oy - qz * oz);
      float dx = 0.5f * (w * ox + qy * oz - qz * oy);
      float dy = 0.5f * (w * oy + qz * ox - qx * oz);
      float dz = 0.5f * (w * oz + qx * oy - qy * ox);
      w += dt * dw; qx += dt * dx; qy += dt * dy; qz += dt * dz;
      float n = sqrtf(fmaxf(w * w + qx * qx + qy * qy + qz * qz, 1e-16f));
      quat[0] = w / n; quat[1] = qx / n; quat[2] = qy / n; quat[3] = qz / n;
    }
    float torso_pen = fmaxf(0.12f - pos[2], 0.0f);
    linvel[2] += dt * KN / TORSO_MASS * torso_pen;
  }

  float forward_vel = (s[0] - x_before) / DT;
  float ctrl_cost = 0.0f;
#pragma unroll
  for (int i = 0; i < ANT_ACT; ++i) ctrl_cost += a[i] * a[i];
  ctrl_cost *= CTRL_COST;
  float cm = contact_mag / SUBSTEPS;
  float contact_cost = CONTACT_COST * cm * cm;
  bool finite = true;
#pragma unroll
  for (int i = 0; i < ANT_STATE; ++i) finite = finite && isfinite(s[i]);
  float z = s[2];
  bool healthy = (z > Z_MIN) && (z < Z_MAX) && finite;
  float r = forward_vel + HEALTHY - ctrl_cost - contact_cost;
  if (!isfinite(r)) r = 0.0f;
  *reward = r;
  *terminated = !healthy;
  if (!finite) {
#pragma unroll
    for (int i = 0; i < ANT_STATE; ++i) s[i] = isfinite(s[i]) ? s[i] : 0.0f;
  }
}
