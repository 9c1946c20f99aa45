// Fused MLP kernels for the Anakin hot path (K2/K3/K6/K7 of SURVEY.md §2.9).
//
// The reference's XLA fuses scan(env.step ∘ net.apply) into few kernels
// (/root/reference/stoix/systems/ppo/anakin/ff_ppo.py:118-146); the eager
// PyTorch equivalent issues ~45 kernels per rollout step. These kernels
// collapse the canonical actor/critic MLP (obs -> Linear(H) -> SiLU ->
// Linear(H) -> SiLU -> heads) into:
//   * policy_value_step_kernel: ONE launch for actor fwd + tanh-normal
//     sample + log-prob + critic fwd over the whole env batch, using
//     v_mfma_f32_16x16x32_bf16 matrix cores with LDS-staged activations;
//   * value_forward_kernel: critic-only fwd (bootstrap values / eval);
//   * ppo_fused_head_loss_kernel: per-row head forward + PPO losses +
//     analytic head backward for the update phase (one wave per sample);
//   * silu fwd/bwd + gather kernels: the epilogue/prologue glue so the
//     update phase is GEMM (hipBLASLt MFMA) + a handful of fused kernels.
//
// Geometry (policy/value forward): one workgroup = 4 waves = 64 batch rows;
// each wave owns a 16-row M-tile and computes every 16-col N-tile of each
// layer with mfma_f32_16x16x32_bf16, staging activations in LDS (row-major,
// +8 col pad -> conflict-free ds_read_b128 A-fragments, §2 of the CDNA4
// guide). Weights are read straight from L2 (nn.Linear row-major [N,K]
// bf16 mirrors; they are hot across the 128 sequential rollout steps).
// Per-wave row blocks are private, so the kernels need NO __syncthreads().
#include "common.h"
#include "ant_core.h"
#include <hip/hip_bf16.h>

typedef __bf16 bf16_t;
typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef __bf16 bf16x4 __attribute__((ext_vector_type(4)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

#define MFMA_BF16_16x16x32 __builtin_amdgcn_mfma_f32_16x16x32_bf16

DEV_INLINE float bf2f(bf16_t x) { return (float)x; }
DEV_INLINE bf16_t f2bf(float x) { return (bf16_t)x; }

DEV_INLINE float silu_f(float x) {
  float s = 1.0f / (1.0f + __expf(-x));
  return x * s;
}
DEV_INLINE float silu_grad_f(float x) {
  float s = 1.0f / (1.0f + __expf(-x));
  return s * (1.0f + x * (1.0f - s));
}
DEV_INLINE float softplus_f(float x) {
  // log1p(exp(x)) stable
  return (x > 20.0f) ? x : log1pf(__expf(x));
}
// Fast-math variants for the per-row loss kernel: built from the
// hardware-accelerated __expf/__logf so the 8-dim unrolled loop pipelines
// instead of serialising on branchy libm calls (atanhf/tanhf/log1pf).
DEV_INLINE float softplus_fast(float x) {
  // |err| vs log1pf path is ~1 ulp of __logf; below x=-15 the result is
  // ~e^x < 3e-7, negligible against min_scale
  return (x > 20.0f) ? x : __logf(1.0f + __expf(x));
}
DEV_INLINE float atanh_fast(float y) {
  // y is pre-clamped to +-(1 - 1e-3)
  return 0.5f * __logf((1.0f + y) / (1.0f - y));
}
DEV_INLINE float tanh_fast(float x) {
  // saturates correctly: __expf overflows to +inf for large 2x -> 1.0
  return 1.0f - 2.0f / (__expf(2.0f * x) + 1.0f);
}

// ---------------------------------------------------------------- probe
// Validates the assumed lane->element mapping of v_mfma_f32_16x16x32_bf16.
// D = A[16,32] @ B[32,16]; two candidate A/B layouts:
//  v0: lane l holds A[m=l&15][k=(l>>4)*8+j], B[k=(l>>4)*8+j][n=l&15]
//  v1: lane l holds A[m=l&15][k=(l>>4)*4+(j&3)+16*(j>>2)] (split-K halves)
// C/D (both): D[row=(l>>4)*4+r][col=l&15] from acc[r].
extern "C" __global__ void mfma_probe_kernel(const bf16_t* __restrict__ A,
                                             const bf16_t* __restrict__ B,
                                             float* __restrict__ D0,
                                             float* __restrict__ D1) {
  int l = threadIdx.x;
  if (l >= 64) return;
  int m = l & 15, g = l >> 4;
  bf16x8 a0, b0, a1, b1;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    int k0 = g * 8 + j;
    a0[j] = A[m * 32 + k0];
    b0[j] = B[k0 * 16 + m];
    int k1 = g * 4 + (j & 3) + 16 * (j >> 2);
    a1[j] = A[m * 32 + k1];
    b1[j] = B[k1 * 16 + m];
  }
  f32x4 z = {0.f, 0.f, 0.f, 0.f};
  f32x4 d0 = MFMA_BF16_16x16x32(a0, b0, z, 0, 0, 0);
  f32x4 d1 = MFMA_BF16_16x16x32(a1, b1, z, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    D0[(g * 4 + r) * 16 + m] = d0[r];
    D1[(g * 4 + r) * 16 + m] = d1[r];
  }
}

// ------------------------------------------------- fused policy/value fwd
//
// LDS: one __shared__ array (guide §5.5 trap 4a), two regions per wave:
//   O: obs tile      [4 waves][16 rows][K1P<=128 + 8 pad] bf16
//   H: activations   [4 waves][16 rows][HID + 8 pad]      bf16
// A-fragment read (ds_read_b128): lane l -> row l&15, k0=(l>>4)*8.
// B-fragment read (global, row-major [N,K] bf16): lane l -> n=nt*16+(l&15),
// k0=ks*32+(l>>4)*8 -> 16B contiguous in K.

#define K1P_MAX 128
#define OPAD 8
#define HPAD 8

// One workgroup = 4 waves sharing ONE 16-row M-tile; each wave owns NT/4
// of the N-tiles per layer (split-N). 4096 envs -> 256 workgroups -> every
// CU busy (the old 64-row-per-WG layout filled only 64 of 256 CUs).
// Buffers: obs tile + per-net ping/pong activation tiles (actor 0/1,
// critic 2/3) in ONE __shared__ struct (guide §5.5 trap 4a).
template <int HID>
struct MlpLds {
  bf16_t O[16][K1P_MAX + OPAD];
  bf16_t H[4][16][HID + HPAD];
  float act[16][8];  // sampled actions for the in-kernel env step
  int done[16];      // per-row episode-end flags from the env phase
  int any_done;      // OR of done[] (gates the bootstrap critic pass)
};

// Load one B fragment from a row-major [N,K] bf16 weight matrix.
template <int HID>
DEV_INLINE bf16x8 load_w_frag(const bf16_t* __restrict__ W, int K, int nt,
                              int ks, int lane) {
  int n = nt * 16 + (lane & 15);
  int k0 = ks * 32 + (lane >> 4) * 8;
  const bf16x8* p = reinterpret_cast<const bf16x8*>(W + (long)n * K + k0);
  return *p;
}

// One MLP layer over the workgroup's 16-row tile, split-N across the 4
// waves: wave `wid` computes N-tiles [wid*NTW, (wid+1)*NTW). Reads A
// fragments from `src` (row-major bf16, stride `sstride`), weights
// W [HID x K] (row-major, L2-hot), bias fp32 [HID]; writes silu(out) to
// `dst`. Caller barriers between dependent layers.
template <int HID, bool ACT_SILU>
DEV_INLINE void wg_layer(const bf16_t* __restrict__ src, int sstride,
                         const bf16_t* __restrict__ W,
                         const float* __restrict__ bias, int K,
                         bf16_t* __restrict__ dst, int dstride, int lane,
                         int wid) {
  constexpr int NTW = HID / 16 / 4;  // N-tiles per wave
  const int KS = K / 32;
  f32x4 acc[NTW];
#pragma unroll
  for (int t = 0; t < NTW; ++t) acc[t] = {0.f, 0.f, 0.f, 0.f};
  const int arow = lane & 15;
  const int ak0 = (lane >> 4) * 8;
  const int nt0 = wid * NTW;
  for (int ks = 0; ks < KS; ++ks) {
    const bf16x8 a =
        *reinterpret_cast<const bf16x8*>(src + arow * sstride + ks * 32 + ak0);
#pragma unroll
    for (int t = 0; t < NTW; ++t) {
      bf16x8 b = load_w_frag<HID>(W, K, nt0 + t, ks, lane);
      acc[t] = MFMA_BF16_16x16x32(a, b, acc[t], 0, 0, 0);
    }
  }
  const int col = lane & 15;
  const int g = lane >> 4;
#pragma unroll
  for (int t = 0; t < NTW; ++t) {
    float b = bias[(nt0 + t) * 16 + col];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float v = acc[t][r] + b;
      if (ACT_SILU) v = silu_f(v);
      dst[(g * 4 + r) * dstride + (nt0 + t) * 16 + col] = f2bf(v);
    }
  }
}

// Stage one wave's 16 obs rows into LDS as bf16 (zero-padded to K1P), with
// optional Welford normalisation (mean/var fp32, reference
// running_statistics.py:205-...); optionally mirror the raw fp32 obs into
// buf_obs[t] (rollout storage) so no separate copy kernel is needed.
DEV_INLINE void stage_obs(const float* __restrict__ obs, int rbase, int OBS,
                          int K1P, bf16_t* __restrict__ Orow, int ostride,
                          const float* __restrict__ nmean,
                          const float* __restrict__ nvar,
                          float* __restrict__ obs_mirror, int tid,
                          int nthreads, int nrows_total) {
  for (int idx = tid; idx < 16 * K1P; idx += nthreads) {
    int r = idx / K1P, k = idx - r * K1P;
    float v = 0.0f;
    if (k < OBS && rbase + r < nrows_total) {
      v = obs[(long)(rbase + r) * OBS + k];
      if (obs_mirror) obs_mirror[(long)(rbase + r) * OBS + k] = v;
      if (nmean) {
        float sd = sqrtf(fmaxf(nvar[k], 1e-6f));
        v = (v - nmean[k]) / sd;
        v = fmaxf(-10.0f, fminf(10.0f, v));
      }
    }
    Orow[r * ostride + k] = f2bf(v);
  }
}

// Critic scalar head via VALU: value[row] = dot(H[row], Wv) + bv.
// 4 lanes per row (lane l: row l>>2, part l&3), butterfly-reduced.
template <int HID>
DEV_INLINE void wave_value_head(const bf16_t* __restrict__ H, int hstride,
                                const bf16_t* __restrict__ Wv, float bv,
                                int rbase, float* __restrict__ value_out,
                                int lane, int B_total) {
  int row = lane >> 2;   // 0..15
  int part = lane & 3;   // 0..3
  float acc = 0.0f;
  for (int c = part * (HID / 4); c < (part + 1) * (HID / 4); c += 8) {
    bf16x8 h = *reinterpret_cast<const bf16x8*>(H + row * hstride + c);
    bf16x8 w = *reinterpret_cast<const bf16x8*>(Wv + c);
#pragma unroll
    for (int j = 0; j < 8; ++j) acc += bf2f(h[j]) * bf2f(w[j]);
  }
  acc += __shfl_xor(acc, 1);
  acc += __shfl_xor(acc, 2);
  if (part == 0 && value_out && rbase + row < B_total)
    value_out[rbase + row] = acc + bv;
}

// The fused policy step. Writes, for rollout step t:
//   buf_obs[t]    = obs                    (fp32 mirror, [B,OBS])
//   action_out    = tanh-normal sample     ([B,ACT] fp32, env input + buffer)
//   logp_out      = log pi(a|s)            ([B] fp32)
//   value_out     = V(s)                   ([B] fp32)
// Head packing: Wh = cat([loc.weight, scale.weight]) [16,HID] bf16,
// bh = cat([loc.bias, scale.bias]) [16] fp32. ACT <= 8.
template <int HID>
__launch_bounds__(256, 2) __global__ void policy_value_step_kernel(
    const float* __restrict__ obs,        // [B, OBS]
    const bf16_t* __restrict__ W1a, const float* __restrict__ b1a,
    const bf16_t* __restrict__ W2a, const float* __restrict__ b2a,
    const bf16_t* __restrict__ Wha, const float* __restrict__ bha,
    const bf16_t* __restrict__ W1c, const float* __restrict__ b1c,
    const bf16_t* __restrict__ W2c, const float* __restrict__ b2c,
    const bf16_t* __restrict__ Wvc, const float* __restrict__ bvc,  // [HID],[1]
    float* __restrict__ obs_mirror,       // [B, OBS] or null
    float* __restrict__ action_out,       // [B, ACT]
    float* __restrict__ logp_out,         // [B]
    float* __restrict__ value_out,        // [B]
    const float* __restrict__ nmean, const float* __restrict__ nvar,
    int B, int OBS, int ACT, float min_scale, float aff_scale,
    float aff_shift, float log_aff_scale, int greedy,
    uint64_t seed, const unsigned int* __restrict__ draw_buf,
    unsigned int draw_offset) {
  __shared__ MlpLds<HID> lds;
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int rbase = blockIdx.x * 16;
  if (rbase >= B) return;
  // draw = base counter + per-graph-node frozen offset: the captured
  // rollout bumps the base ONCE per replay (bump_add) instead of paying a
  // 1-thread bump kernel per step (~5 us each, ~2 ms/update-step)
  const uint32_t draw = (draw_buf ? *draw_buf : 0u) + draw_offset;
  const int K1P = (OBS + 31) & ~31;

  bf16_t* O = &lds.O[0][0];
  constexpr int OS = K1P_MAX + OPAD;
  constexpr int HS = HID + HPAD;
  bf16_t* Ha0 = &lds.H[0][0][0];
  bf16_t* Ha1 = &lds.H[1][0][0];
  bf16_t* Hc0 = &lds.H[2][0][0];
  bf16_t* Hc1 = &lds.H[3][0][0];

  stage_obs(obs, rbase, OBS, K1P, O, OS, nmean, nvar, obs_mirror,
            threadIdx.x, 256, B);
  __syncthreads();

  // ---- torsos (split-N across waves; ping/pong LDS buffers)
  wg_layer<HID, true>(O, OS, W1a, b1a, K1P, Ha0, HS, lane, wid);
  wg_layer<HID, true>(O, OS, W1c, b1c, K1P, Hc0, HS, lane, wid);
  __syncthreads();
  wg_layer<HID, true>(Ha0, HS, W2a, b2a, HID, Ha1, HS, lane, wid);
  wg_layer<HID, true>(Hc0, HS, W2c, b2c, HID, Hc1, HS, lane, wid);
  __syncthreads();

  // ---- critic scalar head on wave 1 (VALU dot)
  if (wid == 1) {
    wave_value_head<HID>(Hc1, HS, Wvc, bvc ? *bvc : 0.0f, rbase, value_out,
                         lane, B);
  }

  // ---- actor head + tanh-normal sample on wave 0:
  // one 16-col N-tile = [loc(0:ACT) | pad | scale(8:8+ACT)]
  if (wid == 0) {
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    const int arow = lane & 15, ak0 = (lane >> 4) * 8;
    for (int ks = 0; ks < HID / 32; ++ks) {
      const bf16x8 a =
          *reinterpret_cast<const bf16x8*>(Ha1 + arow * HS + ks * 32 + ak0);
      bf16x8 b = load_w_frag<HID>(Wha, HID, 0, ks, lane);
      acc = MFMA_BF16_16x16x32(a, b, acc, 0, 0, 0);
    }
    const int col = lane & 15;  // col<8: loc dim, col>=8: scale dim col-8
    const int g = lane >> 4;
    float bh = bha[col];
    float out[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) out[r] = acc[r] + bh;
    // pair loc (lane c) with scale (lane c+8): shfl within the 16-lane group
    const int src_lane = (lane & 48) | (((lane & 15) + 8) & 15);
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float loc = out[r];
      float spre = __shfl(out[r], src_lane, 64);
      float logp = 0.0f;
      float a_val = 0.0f;
      if (col < 8) {
        float sigma = softplus_fast(spre) + min_scale;
        // 4 normals for this (row-block, dim) from one philox block
        int rowblk = (rbase + g * 4) >> 2;  // global row / 4
        Rng4 u = philox_uniform4(seed, 2u, (uint32_t)(rowblk * 16 + col), draw);
        float n[4];
        box_muller(u.a, u.b, &n[0], &n[1]);
        box_muller(u.c, u.d, &n[2], &n[3]);
        float eps = greedy ? 0.0f : n[r];
        float uu = loc + sigma * eps;
        a_val = tanh_fast(uu) * aff_scale + aff_shift;
        // log N(u;loc,sigma) - log|d a/d u|
        float log_det =
            2.0f * (0.6931471805599453f - uu - softplus_fast(-2.0f * uu)) +
            log_aff_scale;
        logp = -0.5f * eps * eps - __logf(sigma) -
               0.9189385332046727f - log_det;
      }
      if (col >= ACT && col < 8) logp = 0.0f;  // unused dims (ACT<8)
      // reduce logp over dims (lanes col 0..7 of this group)
      logp += __shfl_xor(logp, 1);
      logp += __shfl_xor(logp, 2);
      logp += __shfl_xor(logp, 4);
      int grow = rbase + g * 4 + r;
      if (grow < B) {
        if (col < ACT) action_out[(long)grow * ACT + col] = a_val;
        if (col == 0) logp_out[grow] = logp;
      }
    }
  }
}

// Critic-only forward (bootstrap values: V(extras["next_obs"]),
// ff_ppo.py:113-116).
template <int HID>
__launch_bounds__(256, 2) __global__ void value_forward_kernel(
    const float* __restrict__ obs, const bf16_t* __restrict__ W1c,
    const float* __restrict__ b1c, const bf16_t* __restrict__ W2c,
    const float* __restrict__ b2c, const bf16_t* __restrict__ Wvc,
    const float* __restrict__ bvc, float* __restrict__ value_out,
    const float* __restrict__ nmean, const float* __restrict__ nvar, int B,
    int OBS) {
  __shared__ MlpLds<HID> lds;
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int rbase = blockIdx.x * 16;
  if (rbase >= B) return;
  const int K1P = (OBS + 31) & ~31;
  bf16_t* O = &lds.O[0][0];
  bf16_t* H0 = &lds.H[0][0][0];
  bf16_t* H1 = &lds.H[1][0][0];
  constexpr int OS = K1P_MAX + OPAD;
  constexpr int HS = HID + HPAD;
  stage_obs(obs, rbase, OBS, K1P, O, OS, nmean, nvar, nullptr, threadIdx.x,
            256, B);
  __syncthreads();
  wg_layer<HID, true>(O, OS, W1c, b1c, K1P, H0, HS, lane, wid);
  __syncthreads();
  wg_layer<HID, true>(H0, HS, W2c, b2c, HID, H1, HS, lane, wid);
  __syncthreads();
  if (wid == 0) {
    wave_value_head<HID>(H1, HS, Wvc, bvc ? *bvc : 0.0f, rbase, value_out,
                         lane, B);
  }
}

// --------------------------------------- fused rollout step (Ant)
//
// ONE launch per rollout step: actor fwd + tanh-normal sample + critic fwd
// (MFMA, all 4 waves) -> Ant physics + autoreset + episode metrics (16 env
// threads of the workgroup; everything is per-env-row independent, so no
// grid-wide sync is needed) -> critic fwd on the pre-reset next_obs
// (bootstrap). Replaces the 3-kernel (policy/env/value) chain: fewer
// launches, 4x more CUs on the physics (256 WGs vs 64), and next_obs never
// round-trips through HBM (it stays in LDS for the bootstrap pass).
// RPW = batch rows (envs) per workgroup (16 = full MFMA M-tile; the 8-row
// variant exists for the occupancy experiment documented in the launcher
// -- it lost, because a half-empty tile still issues the full instruction
// stream).
template <int HID, int RPW>
__launch_bounds__(256, 2) __global__ void rollout_step_ant_kernel(
    float* __restrict__ obs_io,          // [B, 27] env obs buffer (in/out)
    float* __restrict__ env_state,       // [B, 29]
    int* __restrict__ step_count, float* __restrict__ ep_return,
    int* __restrict__ ep_length, float* __restrict__ last_ep_return,
    int* __restrict__ last_ep_length,
    const bf16_t* __restrict__ W1a, const float* __restrict__ b1a,
    const bf16_t* __restrict__ W2a, const float* __restrict__ b2a,
    const bf16_t* __restrict__ Wha, const float* __restrict__ bha,
    const bf16_t* __restrict__ W1c, const float* __restrict__ b1c,
    const bf16_t* __restrict__ W2c, const float* __restrict__ b2c,
    const bf16_t* __restrict__ Wvc, const float* __restrict__ bvc,
    float* __restrict__ buf_obs,         // [B, 27] rollout storage row t
    float* __restrict__ buf_action,      // [B, 8]
    float* __restrict__ buf_logp,        // [B]
    float* __restrict__ buf_value,       // [B]
    float* __restrict__ buf_bootstrap,   // [B] V(next_obs), DONE rows only
    float* __restrict__ buf_reward, float* __restrict__ buf_discount,
    unsigned char* __restrict__ buf_steptype,
    int B, int OBS, int ACT, int max_episode_steps, float min_scale,
    float aff_scale, float aff_shift, float log_aff_scale,
    uint64_t policy_seed, uint64_t env_seed,
    const unsigned int* __restrict__ policy_draw,
    const unsigned int* __restrict__ env_draw, unsigned int draw_offset) {
  __shared__ MlpLds<HID> lds;
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int rbase = blockIdx.x * RPW;
  if (rbase >= B) return;
  // rows [RPW, 16) of the tile belong to the NEXT workgroup at RPW=8:
  // cap every per-row bound so they are staged as zeros and never written
  const int Bcap = min(B, rbase + RPW);
  const uint32_t pdraw = *policy_draw + draw_offset;
  const uint32_t edraw = *env_draw + draw_offset;
  const int K1P = (OBS + 31) & ~31;

  bf16_t* O = &lds.O[0][0];
  constexpr int OS = K1P_MAX + OPAD;
  constexpr int HS = HID + HPAD;
  bf16_t* Ha0 = &lds.H[0][0][0];
  bf16_t* Ha1 = &lds.H[1][0][0];
  bf16_t* Hc0 = &lds.H[2][0][0];
  bf16_t* Hc1 = &lds.H[3][0][0];

  if (threadIdx.x == 0) lds.any_done = 0;
  stage_obs(obs_io, rbase, OBS, K1P, O, OS, nullptr, nullptr, buf_obs,
            threadIdx.x, 256, Bcap);
  __syncthreads();

  // ---- policy + value (same structure as policy_value_step_kernel)
  wg_layer<HID, true>(O, OS, W1a, b1a, K1P, Ha0, HS, lane, wid);
  wg_layer<HID, true>(O, OS, W1c, b1c, K1P, Hc0, HS, lane, wid);
  __syncthreads();
  wg_layer<HID, true>(Ha0, HS, W2a, b2a, HID, Ha1, HS, lane, wid);
  wg_layer<HID, true>(Hc0, HS, W2c, b2c, HID, Hc1, HS, lane, wid);
  __syncthreads();

  if (wid == 1) {
    wave_value_head<HID>(Hc1, HS, Wvc, bvc ? *bvc : 0.0f, rbase, buf_value,
                         lane, Bcap);
  }
  if (wid == 0) {
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    const int arow = lane & 15, ak0 = (lane >> 4) * 8;
    for (int ks = 0; ks < HID / 32; ++ks) {
      const bf16x8 aa =
          *reinterpret_cast<const bf16x8*>(Ha1 + arow * HS + ks * 32 + ak0);
      bf16x8 bb = load_w_frag<HID>(Wha, HID, 0, ks, lane);
      acc = MFMA_BF16_16x16x32(aa, bb, acc, 0, 0, 0);
    }
    const int col = lane & 15;
    const int g = lane >> 4;
    float bh = bha[col];
    float out[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) out[r] = acc[r] + bh;
    const int src_lane = (lane & 48) | (((lane & 15) + 8) & 15);
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float loc = out[r];
      float spre = __shfl(out[r], src_lane, 64);
      float logp = 0.0f;
      float a_val = 0.0f;
      if (col < 8) {
        float sigma = softplus_fast(spre) + min_scale;
        int rowblk = (rbase + g * 4) >> 2;
        Rng4 u = philox_uniform4(policy_seed, 2u, (uint32_t)(rowblk * 16 + col),
                                 pdraw);
        float n[4];
        box_muller(u.a, u.b, &n[0], &n[1]);
        box_muller(u.c, u.d, &n[2], &n[3]);
        float eps = n[r];
        float uu = loc + sigma * eps;
        a_val = tanh_fast(uu) * aff_scale + aff_shift;
        float log_det =
            2.0f * (0.6931471805599453f - uu - softplus_fast(-2.0f * uu)) +
            log_aff_scale;
        logp = -0.5f * eps * eps - __logf(sigma) - 0.9189385332046727f -
               log_det;
      }
      if (col >= ACT && col < 8) logp = 0.0f;
      logp += __shfl_xor(logp, 1);
      logp += __shfl_xor(logp, 2);
      logp += __shfl_xor(logp, 4);
      int grow = rbase + g * 4 + r;
      if (grow < Bcap) {
        if (col < ACT) {
          buf_action[(long)grow * ACT + col] = a_val;
          lds.act[g * 4 + r][col] = a_val;
        }
        if (col == 0) buf_logp[grow] = logp;
      }
    }
  }
  __syncthreads();

  // ---- env step: wave 0's 64 lanes, FOUR lanes per env (one leg each;
  // ant_physics_step_x4 quad-cooperative physics)
  if (threadIdx.x < 64) {
    int row = threadIdx.x >> 2;
    int leg = threadIdx.x & 3;
    int b = rbase + row;
    if (b < Bcap) {
      float es[ANT_STATE];
#pragma unroll
      for (int i = 0; i < ANT_STATE; ++i) es[i] = env_state[b * ANT_STATE + i];
      float ea[ANT_ACT];
#pragma unroll
      for (int i = 0; i < ANT_ACT; ++i)
        ea[i] = fminf(fmaxf(lds.act[row][i], -1.0f), 1.0f);
      float reward;
      bool terminated;
      ant_physics_step_x4(es, ea, leg, &reward, &terminated);
      if (leg == 0) {  // quad lane 0 does all the bookkeeping/writes
      int sc = step_count[b] + 1;
      bool truncated = (sc >= max_episode_steps) && !terminated;
      bool done = terminated || truncated;
      float ret = ep_return[b] + reward;
      int len = ep_length[b] + 1;
      if (done) { last_ep_return[b] = ret; last_ep_length[b] = len; }

      // pre-reset next_obs -> LDS obs tile; the in-kernel bootstrap
      // critic below runs ONLY for workgroups with done rows: for every
      // non-done row V(next_obs) == V(obs_{t+1}) == next step's buf_value,
      // which the engine fills in with one shifted masked copy
      {
        float nobs[ANT_OBS];
        ant_write_obs(es, nobs);
#pragma unroll
        for (int k = 0; k < ANT_OBS; ++k) lds.O[row][k] = f2bf(nobs[k]);
        lds.done[row] = done ? 1 : 0;
        if (done) lds.any_done = 1;
      }

      if (done) {
        ant_reset_state(es, env_seed, (uint32_t)b, edraw);
        sc = 0; ret = 0.0f; len = 0;
      }
#pragma unroll
      for (int i = 0; i < ANT_STATE; ++i) env_state[b * ANT_STATE + i] = es[i];
      ant_write_obs(es, obs_io + b * ANT_OBS);  // post-reset obs for t+1
      step_count[b] = sc;
      ep_return[b] = ret;
      ep_length[b] = len;
      buf_reward[b] = reward;
      buf_discount[b] = terminated ? 0.0f : 1.0f;
      buf_steptype[b] =
          terminated ? ST_TERMINATED : (truncated ? ST_TRUNCATED : ST_MID);
      }  // leg == 0
    }
  }
  __syncthreads();

  // ---- bootstrap critic, only when this WG saw an episode end
  if (lds.any_done) {
    wg_layer<HID, true>(O, OS, W1c, b1c, K1P, Hc0, HS, lane, wid);
    __syncthreads();
    wg_layer<HID, true>(Hc0, HS, W2c, b2c, HID, Hc1, HS, lane, wid);
    __syncthreads();
    if (wid == 0) {
      // value head, but store only the done rows (others get the shifted
      // next-step value from the engine)
      int row = lane >> 2;
      int part = lane & 3;
      float acc = 0.0f;
      for (int c = part * (HID / 4); c < (part + 1) * (HID / 4); c += 8) {
        bf16x8 h = *reinterpret_cast<const bf16x8*>(Hc1 + row * HS + c);
        bf16x8 w = *reinterpret_cast<const bf16x8*>(Wvc + c);
#pragma unroll
        for (int j = 0; j < 8; ++j) acc += bf2f(h[j]) * bf2f(w[j]);
      }
      acc += __shfl_xor(acc, 1);
      acc += __shfl_xor(acc, 2);
      if (part == 0 && rbase + row < Bcap && lds.done[row])
        buf_bootstrap[rbase + row] = acc + (bvc ? *bvc : 0.0f);
    }
  }
}

extern "C" void launch_rollout_step_ant(
    float* obs_io, float* env_state, int* step_count, float* ep_return,
    int* ep_length, float* last_ep_return, int* last_ep_length,
    const void* W1a, const float* b1a, const void* W2a, const float* b2a,
    const void* Wha, const float* bha, const void* W1c, const float* b1c,
    const void* W2c, const float* b2c, const void* Wvc, const float* bvc,
    float* buf_obs, float* buf_action, float* buf_logp, float* buf_value,
    float* buf_bootstrap, float* buf_reward, float* buf_discount,
    unsigned char* buf_steptype, int B, int OBS, int ACT, int HID,
    int max_episode_steps, float min_scale, float aff_scale, float aff_shift,
    float log_aff_scale, uint64_t policy_seed, uint64_t env_seed,
    unsigned int* policy_draw, unsigned int* env_draw,
    unsigned int draw_offset, void* stream) {
  hipStream_t s = (hipStream_t)stream;
  // RPW=8 was tried to double the grid for stall overlap at B=4096
  // (1 WG/CU): measured WORSE (rollout 4.0 -> 5.4 ms) because each WG
  // still issues the full 16-row tile instruction stream -- halving the
  // rows doubles total issue. 16 stays.
  int rpw = 16;
  dim3 grid((B + rpw - 1) / rpw), block(256);
  if (HID == 256 && rpw == 8) {
    hipLaunchKernelGGL((rollout_step_ant_kernel<256, 8>), grid, block, 0, s,
                       obs_io, env_state, step_count, ep_return, ep_length,
                       last_ep_return, last_ep_length, (const bf16_t*)W1a,
                       b1a, (const bf16_t*)W2a, b2a, (const bf16_t*)Wha, bha,
                       (const bf16_t*)W1c, b1c, (const bf16_t*)W2c, b2c,
                       (const bf16_t*)Wvc, bvc, buf_obs, buf_action, buf_logp,
                       buf_value, buf_bootstrap, buf_reward, buf_discount,
                       buf_steptype, B, OBS, ACT, max_episode_steps,
                       min_scale, aff_scale, aff_shift, log_aff_scale,
                       policy_seed, env_seed, policy_draw, env_draw,
                       draw_offset);
  } else if (HID == 256) {
    hipLaunchKernelGGL((rollout_step_ant_kernel<256, 16>), grid, block, 0, s,
                       obs_io, env_state, step_count, ep_return, ep_length,
                       last_ep_return, last_ep_length, (const bf16_t*)W1a,
                       b1a, (const bf16_t*)W2a, b2a, (const bf16_t*)Wha, bha,
                       (const bf16_t*)W1c, b1c, (const bf16_t*)W2c, b2c,
                       (const bf16_t*)Wvc, bvc, buf_obs, buf_action, buf_logp,
                       buf_value, buf_bootstrap, buf_reward, buf_discount,
                       buf_steptype, B, OBS, ACT, max_episode_steps,
                       min_scale, aff_scale, aff_shift, log_aff_scale,
                       policy_seed, env_seed, policy_draw, env_draw,
                       draw_offset);
  } else if (rpw == 8) {
    hipLaunchKernelGGL((rollout_step_ant_kernel<128, 8>), grid, block, 0, s,
                       obs_io, env_state, step_count, ep_return, ep_length,
                       last_ep_return, last_ep_length, (const bf16_t*)W1a,
                       b1a, (const bf16_t*)W2a, b2a, (const bf16_t*)Wha, bha,
                       (const bf16_t*)W1c, b1c, (const bf16_t*)W2c, b2c,
                       (const bf16_t*)Wvc, bvc, buf_obs, buf_action, buf_logp,
                       buf_value, buf_bootstrap, buf_reward, buf_discount,
                       buf_steptype, B, OBS, ACT, max_episode_steps,
                       min_scale, aff_scale, aff_shift, log_aff_scale,
                       policy_seed, env_seed, policy_draw, env_draw,
                       draw_offset);
  } else {
    hipLaunchKernelGGL((rollout_step_ant_kernel<128, 16>), grid, block, 0, s,
                       obs_io, env_state, step_count, ep_return, ep_length,
                       last_ep_return, last_ep_length, (const bf16_t*)W1a,
                       b1a, (const bf16_t*)W2a, b2a, (const bf16_t*)Wha, bha,
                       (const bf16_t*)W1c, b1c, (const bf16_t*)W2c, b2c,
                       (const bf16_t*)Wvc, bvc, buf_obs, buf_action, buf_logp,
                       buf_value, buf_bootstrap, buf_reward, buf_discount,
                       buf_steptype, B, OBS, ACT, max_episode_steps,
                       min_scale, aff_scale, aff_shift, log_aff_scale,
                       policy_seed, env_seed, policy_draw, env_draw,
                       draw_offset);
  }
}

// ------------------------------------ fused Linear+SiLU forward (update)
//
// Z = X @ W^T + b, H = silu(Z), both stored bf16. Replaces a hipBLASLt
// GemmAndBias (~19 us at M=32768,N=256,K=256) + a separate silu kernel
// (~7 us) per layer. Tile: 128 rows x 128 cols per WG (4 waves as 2x2,
// 64x64 each); X staged in LDS double-buffered, W fragments straight from
// L2 (row-major [N,K] bf16 mirrors, hot across minibatches).
#define LSF_BM 128
#define LSF_BK 32
#define LSF_APAD 8

struct LinSiluLds {
  bf16_t A[2][LSF_BM][LSF_BK + LSF_APAD];
};

template <bool SILU>
__launch_bounds__(256, 2) __global__ void linear_silu_kernel(
    const bf16_t* __restrict__ X,  // [S, K]
    const bf16_t* __restrict__ W,  // [N, K] row-major
    const float* __restrict__ bias,  // [N] fp32 (master views)
    bf16_t* __restrict__ Z,        // [S, N]
    bf16_t* __restrict__ H,        // [S, N] (silu(Z)) or null
    int S, int K, int N) {
  __shared__ LinSiluLds lds;
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int m0 = blockIdx.x * LSF_BM;
  const int n_blk = blockIdx.y * 128;
  // wave -> 64x64 quadrant
  const int wm = (wid >> 1) * 64;  // 0 or 64 (row offset in tile)
  const int wn = (wid & 1) * 64;   // 0 or 64 (col offset)
  const int KS = K / LSF_BK;

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  // stage k-step `ks` of the X tile into buffer pp: 128x32 bf16 = 8 KB,
  // 256 threads x 2 16B pieces, fully coalesced both sides
  auto stage = [&](int ks, int pp) {
#pragma unroll
    for (int q = 0; q < 2; ++q) {
      int flat = q * 256 + threadIdx.x;  // 512 chunks of 8 elems
      int row = flat >> 2;
      int kh = (flat & 3) * 8;
      *reinterpret_cast<bf16x8*>(&lds.A[pp][row][kh]) =
          *reinterpret_cast<const bf16x8*>(X + (long)(m0 + row) * K +
                                           ks * LSF_BK + kh);
    }
  };

  stage(0, 0);
  const int arow_base = wm + (lane & 15);
  const int ak0 = (lane >> 4) * 8;
  for (int ks = 0; ks < KS; ++ks) {
    __syncthreads();  // staged tile visible to all waves
    if (ks + 1 < KS) stage(ks + 1, (ks + 1) & 1);
    // B fragments hoisted out of the mi loop (they depend only on nj, ks;
    // reloading them per mi quadrupled the L2 W-traffic)
    bf16x8 b[4];
#pragma unroll
    for (int nj = 0; nj < 4; ++nj) {
      int n = n_blk + wn + nj * 16 + (lane & 15);
      b[nj] = *reinterpret_cast<const bf16x8*>(W + (long)n * K +
                                               ks * LSF_BK + ak0);
    }
#pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
      const bf16x8 a = *reinterpret_cast<const bf16x8*>(
          &lds.A[ks & 1][arow_base + mi * 16][ak0]);
#pragma unroll
      for (int nj = 0; nj < 4; ++nj) {
        acc[mi][nj] = MFMA_BF16_16x16x32(a, b[nj], acc[mi][nj], 0, 0, 0);
      }
    }
  }

  // epilogue: bias + (silu) -> Z, H
  const int col_l = lane & 15;
  const int g = lane >> 4;
#pragma unroll
  for (int nj = 0; nj < 4; ++nj) {
    int n = n_blk + wn + nj * 16 + col_l;
    float bv = bias[n];
#pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int m = m0 + wm + mi * 16 + g * 4 + r;
        float z = acc[mi][nj][r] + bv;
        Z[(long)m * N + n] = f2bf(z);
        if (SILU && H) H[(long)m * N + n] = f2bf(silu_f(z));
      }
    }
  }
}

extern "C" void launch_linear_silu(const void* X, const void* W,
                                   const float* bias, void* Z, void* H,
                                   int S, int K, int N, int do_silu,
                                   void* stream) {
  dim3 grid(S / LSF_BM, N / 128), block(256);
  if (do_silu) {
    hipLaunchKernelGGL(linear_silu_kernel<true>, grid, block, 0,
                       (hipStream_t)stream, (const bf16_t*)X,
                       (const bf16_t*)W, bias, (bf16_t*)Z, (bf16_t*)H, S, K,
                       N);
  } else {
    hipLaunchKernelGGL(linear_silu_kernel<false>, grid, block, 0,
                       (hipStream_t)stream, (const bf16_t*)X,
                       (const bf16_t*)W, bias, (bf16_t*)Z, (bf16_t*)H, S, K,
                       N);
  }
}

// ------------------------------------------------------- update-phase glue

// silu fwd/bwd, bf16, 8-wide vectorised (guide common-mistake #2).
extern "C" __global__ void silu_fwd_kernel(const bf16_t* __restrict__ z,
                                           bf16_t* __restrict__ h, long n) {
  long i = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  long stride = (long)gridDim.x * blockDim.x * 8;
  for (; i < n; i += stride) {
    bf16x8 zv = *reinterpret_cast<const bf16x8*>(z + i);
    bf16x8 hv;
#pragma unroll
    for (int j = 0; j < 8; ++j) hv[j] = f2bf(silu_f(bf2f(zv[j])));
    *reinterpret_cast<bf16x8*>(h + i) = hv;
  }
}

extern "C" __global__ void silu_bwd_kernel(const bf16_t* __restrict__ dh,
                                           const bf16_t* __restrict__ z,
                                           bf16_t* __restrict__ dz, long n) {
  long i = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  long stride = (long)gridDim.x * blockDim.x * 8;
  for (; i < n; i += stride) {
    bf16x8 dv = *reinterpret_cast<const bf16x8*>(dh + i);
    bf16x8 zv = *reinterpret_cast<const bf16x8*>(z + i);
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      o[j] = f2bf(bf2f(dv[j]) * silu_grad_f(bf2f(zv[j])));
    *reinterpret_cast<bf16x8*>(dz + i) = o;
  }
}

// Fused minibatch gather: one kernel replaces 6 index_selects. Gathers the
// permuted rows of the flat rollout storage for one minibatch (obs -> bf16
// GEMM input; the scalar fields fp32).
extern "C" __global__ void ppo_gather_kernel(
    const long* __restrict__ idx, int mb_size,
    const float* __restrict__ obs, int OBS, int OBS_PAD,
    const float* __restrict__ action, int ACT,
    const float* __restrict__ logp, const float* __restrict__ value,
    const float* __restrict__ adv, const float* __restrict__ targets,
    bf16_t* __restrict__ obs_out, float* __restrict__ action_out,
    float* __restrict__ logp_out, float* __restrict__ value_out,
    float* __restrict__ adv_out, float* __restrict__ targets_out,
    const float* __restrict__ nmean, const float* __restrict__ nvar) {
  // one wave per row; obs_out is [mb, OBS_PAD] (K padded to the MFMA
  // K-step, pad columns zero-filled to match the padded W1 layout)
  int row = (blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  int lane = threadIdx.x & 63;
  if (row >= mb_size) return;
  long src = idx[row];
  for (int k = lane; k < OBS_PAD; k += 64) {
    float v = 0.0f;
    if (k < OBS) {
      v = obs[src * OBS + k];
      if (nmean) {
        float sd = sqrtf(fmaxf(nvar[k], 1e-6f));
        v = fmaxf(-10.0f, fminf(10.0f, (v - nmean[k]) / sd));
      }
    }
    obs_out[(long)row * OBS_PAD + k] = f2bf(v);
  }
  if (lane < ACT) action_out[(long)row * ACT + lane] = action[src * ACT + lane];
  if (lane == 0) {
    logp_out[row] = logp[src];
    value_out[row] = value[src];
    adv_out[row] = adv[src];
    targets_out[row] = targets[src];
  }
}

// ---------------------------------------------- fused PPO head + losses
//
// One THREAD per sample row. The head projections themselves are GEMMs and
// run on hipBLASLt in the engine (heads = H2a @ Wha^T + bha -> [B,16],
// v = H2c @ Wvc + bvc -> [B]); this kernel does the remaining per-row
// scalar math: tanh-normal log-prob of the stored action, PPO clip loss
// (losses.py ppo_clip_loss), clipped value loss, MC entropy, and the
// analytic gradients d(total)/d{loc, scale_pre, v} (verified against
// autograd in tests/test_fused_math.py). dH2a/dH2c are then GEMMs again
// (dhead @ Wha, dv @ Wvc) in the engine.
//
// Gradient scale: total = a_loss - ent_coef*entropy + vf_coef*v_loss,
// all means over the minibatch (inv_B folded in here).
extern "C" __global__ void ppo_head_loss_kernel(
    const bf16_t* __restrict__ heads,    // [B, 16] = loc(0:8)|spre(8:16)
    const bf16_t* __restrict__ v_in,     // [B] critic head output
    const float* __restrict__ action,    // [B, ACT]
    const float* __restrict__ old_logp,  // [B]
    const float* __restrict__ old_value, // [B]
    const float* __restrict__ adv,       // [B]
    const float* __restrict__ targets,   // [B]
    bf16_t* __restrict__ dhead,  // [B, 16]
    bf16_t* __restrict__ dv_out, // [B] (for the dH2c outer-product GEMM)
    bf16_t* __restrict__ dv16_out,  // [B,16] col 0 (wgrad A-operand) or null
    float* __restrict__ metrics, // [3]: actor_loss, value_loss, entropy
    int B, int ACT, float clip_eps, float ent_coef, float vf_coef,
    float min_scale, float aff_scale, float aff_shift, float log_aff_scale,
    float inv_B, uint64_t seed, const unsigned int* __restrict__ draw_buf,
    unsigned int draw_offset) {
  const int row = blockIdx.x * blockDim.x + threadIdx.x;
  const bool active = row < B;  // inactive lanes still join the metric
                                // shuffle-reduce (contribute zeros)
  const uint32_t draw = (draw_buf ? *draw_buf : 0u) + draw_offset;

  // ---- per-dim tanh-normal forward
  float loc[8], spre[8], sigma[8], u[8], eps_e[8], u_e[8];
  float a_row[8];
  if (active) {
    const bf16x8* h8 = reinterpret_cast<const bf16x8*>(heads + (long)row * 16);
    bf16x8 hl = h8[0], hs = h8[1];
    // issue the action-row loads early (two float4 vectors when ACT fills
    // them); at 0.5 resident waves/SIMD there is no other wave to hide
    // their latency, so overlap them with the Philox/softplus ALU below
    const float* arow_p = action + (long)row * ACT;
#pragma unroll
    for (int j4 = 0; j4 < 8; j4 += 4) {
      if (j4 < ACT && (ACT & 3) == 0) {
        const f32x4 av = *reinterpret_cast<const f32x4*>(arow_p + j4);
        a_row[j4 + 0] = av[0]; a_row[j4 + 1] = av[1];
        a_row[j4 + 2] = av[2]; a_row[j4 + 3] = av[3];
      }
    }
    if ((ACT & 3) != 0) {
      for (int j = 0; j < ACT; ++j) a_row[j] = arow_p[j];
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      loc[j] = bf2f(hl[j]);
      spre[j] = bf2f(hs[j]);
      sigma[j] = softplus_fast(spre[j]) + min_scale;
    }
  }
  float logp_new = 0.0f, ent = 0.0f;
  // one Philox call serves 4 dims (4 uniforms -> 4 normals via two
  // box_muller pairs) instead of one call per dim discarding half its draws
  float nrm[8];
#pragma unroll
  for (int q = 0; q < 2; ++q) {
    if (q * 4 >= ACT || !active) break;
    Rng4 uu = philox_uniform4(seed, 3u, (uint32_t)(row * 2 + q), draw);
    box_muller(uu.a, uu.b, &nrm[q * 4 + 0], &nrm[q * 4 + 1]);
    box_muller(uu.c, uu.d, &nrm[q * 4 + 2], &nrm[q * 4 + 3]);
  }
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    if (j >= ACT || !active) break;
    float a = a_row[j];
    float y = (a - aff_shift) / aff_scale;
    y = fmaxf(-1.0f + 1e-3f, fminf(1.0f - 1e-3f, y));
    u[j] = atanh_fast(y);
    float z = (u[j] - loc[j]) / sigma[j];
    float log_sigma = __logf(sigma[j]);
    float log_det =
        2.0f * (0.6931471805599453f - u[j] - softplus_fast(-2.0f * u[j])) +
        log_aff_scale;
    logp_new +=
        -0.5f * z * z - log_sigma - 0.9189385332046727f - log_det;
    // MC entropy sample: u' = loc + sigma*eps'
    float n1 = nrm[j];
    eps_e[j] = n1;
    u_e[j] = loc[j] + sigma[j] * n1;
    float log_det_e =
        2.0f * (0.6931471805599453f - u_e[j] - softplus_fast(-2.0f * u_e[j])) +
        log_aff_scale;
    ent -= -0.5f * n1 * n1 - log_sigma - 0.9189385332046727f -
           log_det_e;
  }

  // ---- PPO clip loss
  float A = active ? adv[row] : 0.0f;
  float ratio = __expf(logp_new - (active ? old_logp[row] : 0.0f));
  float r_clip = fmaxf(1.0f - clip_eps, fminf(1.0f + clip_eps, ratio));
  float l1 = ratio * A, l2 = r_clip * A;
  float a_loss = -fminf(l1, l2);
  float dl_dlogp;  // autograd semantics: min picks l1 branch on tie
  if (l1 <= l2) {
    dl_dlogp = -ratio * A;
  } else {
    dl_dlogp = (ratio > 1.0f - clip_eps && ratio < 1.0f + clip_eps)
                   ? -ratio * A
                   : 0.0f;
  }
  dl_dlogp *= inv_B;

  // ---- clipped value loss
  float v_pred = active ? bf2f(v_in[row]) : 0.0f;
  float ov = active ? old_value[row] : 0.0f;
  float tg = active ? targets[row] : 0.0f;
  float v_clip = ov + fmaxf(-clip_eps, fminf(clip_eps, v_pred - ov));
  float e1 = (v_pred - tg), e2 = (v_clip - tg);
  float sq1 = e1 * e1, sq2 = e2 * e2;
  float v_loss = 0.5f * fmaxf(sq1, sq2);
  float dv;
  if (sq1 >= sq2) {
    dv = e1;
  } else {
    dv = (fabsf(v_pred - ov) < clip_eps) ? e2 : 0.0f;
  }
  dv *= vf_coef * inv_B;

  // ---- analytic head gradients
  bf16x8 dl8, ds8;
  const float ce = -ent_coef * inv_B;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    float dloc = 0.0f, dspre = 0.0f;
    if (j < ACT) {
      float z = (u[j] - loc[j]) / sigma[j];
      dloc = dl_dlogp * (z / sigma[j]);
      float dsig = dl_dlogp * ((z * z - 1.0f) / sigma[j]);
      float th = tanh_fast(u_e[j]);
      dloc += ce * (-2.0f * th);
      dsig += ce * (1.0f / sigma[j] - 2.0f * th * eps_e[j]);
      dspre = dsig * (1.0f / (1.0f + __expf(-spre[j])));
    }
    dl8[j] = f2bf(dloc);
    ds8[j] = f2bf(dspre);
  }
  if (active) {
    bf16x8* out8 = reinterpret_cast<bf16x8*>(dhead + (long)row * 16);
    out8[0] = dl8;
    out8[1] = ds8;
    dv_out[row] = f2bf(dv);
    if (dv16_out) dv16_out[(long)row * 16] = f2bf(dv);
  }

  // ---- loss metrics (wave-level pre-reduce, one atomic per wave).
  // Optional: 512 waves funnelling atomics into the same 3 words
  // serialise across all XCDs, so the engine requests metrics only on
  // the minibatch it actually reports (the last one).
  if (metrics) {
    float m0 = active ? a_loss * inv_B : 0.0f;
    float m1 = active ? v_loss * inv_B : 0.0f;
    float m2 = active ? ent * inv_B : 0.0f;
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      m0 += __shfl_down(m0, off);
      m1 += __shfl_down(m1, off);
      m2 += __shfl_down(m2, off);
    }
    if ((threadIdx.x & 63) == 0) {
      atomicAdd(&metrics[0], m0);
      atomicAdd(&metrics[1], m1);
      atomicAdd(&metrics[2], m2);
    }
  }
}

// --------------------------------------------------------- host launchers

extern "C" __global__ void bump_u32_kernel2(unsigned int* p) {
  if (blockIdx.x == 0 && threadIdx.x == 0) (*p)++;
}

extern "C" __global__ void bump_add_kernel(unsigned int* p, unsigned int n) {
  if (blockIdx.x == 0 && threadIdx.x == 0) (*p) += n;
}

extern "C" void launch_bump_add(unsigned int* p, unsigned int n,
                                void* stream) {
  hipLaunchKernelGGL(bump_add_kernel, dim3(1), dim3(1), 0,
                     (hipStream_t)stream, p, n);
}

extern "C" void launch_mfma_probe(const void* A, const void* B, float* D0,
                                  float* D1, void* stream) {
  hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0,
                     (hipStream_t)stream, (const bf16_t*)A, (const bf16_t*)B,
                     D0, D1);
}

extern "C" void launch_policy_value_step(
    const float* obs, const void* W1a, const float* b1a, const void* W2a,
    const float* b2a, const void* Wha, const float* bha, const void* W1c,
    const float* b1c, const void* W2c, const float* b2c, const void* Wvc,
    const float* bvc, float* obs_mirror, float* action_out, float* logp_out,
    float* value_out, const float* nmean, const float* nvar, int B, int OBS,
    int ACT, int HID, float min_scale, float aff_scale, float aff_shift,
    float log_aff_scale, int greedy, uint64_t seed, unsigned int* draw_buf,
    unsigned int draw_offset, int do_bump, void* stream) {
  hipStream_t s = (hipStream_t)stream;
  dim3 grid((B + 15) / 16), block(256);
  if (HID == 512) {
    hipLaunchKernelGGL(policy_value_step_kernel<512>, grid, block, 0, s, obs,
                       (const bf16_t*)W1a, b1a, (const bf16_t*)W2a, b2a,
                       (const bf16_t*)Wha, bha, (const bf16_t*)W1c, b1c,
                       (const bf16_t*)W2c, b2c, (const bf16_t*)Wvc, bvc,
                       obs_mirror, action_out, logp_out, value_out, nmean,
                       nvar, B, OBS, ACT, min_scale, aff_scale, aff_shift,
                       log_aff_scale, greedy, seed, draw_buf, draw_offset);
  } else if (HID == 256) {
    hipLaunchKernelGGL(policy_value_step_kernel<256>, grid, block, 0, s, obs,
                       (const bf16_t*)W1a, b1a, (const bf16_t*)W2a, b2a,
                       (const bf16_t*)Wha, bha, (const bf16_t*)W1c, b1c,
                       (const bf16_t*)W2c, b2c, (const bf16_t*)Wvc, bvc,
                       obs_mirror, action_out, logp_out, value_out, nmean,
                       nvar, B, OBS, ACT, min_scale, aff_scale, aff_shift,
                       log_aff_scale, greedy, seed, draw_buf, draw_offset);
  } else {
    hipLaunchKernelGGL(policy_value_step_kernel<128>, grid, block, 0, s, obs,
                       (const bf16_t*)W1a, b1a, (const bf16_t*)W2a, b2a,
                       (const bf16_t*)Wha, bha, (const bf16_t*)W1c, b1c,
                       (const bf16_t*)W2c, b2c, (const bf16_t*)Wvc, bvc,
                       obs_mirror, action_out, logp_out, value_out, nmean,
                       nvar, B, OBS, ACT, min_scale, aff_scale, aff_shift,
                       log_aff_scale, greedy, seed, draw_buf, draw_offset);
  }
  if (draw_buf && do_bump)
    hipLaunchKernelGGL(bump_u32_kernel2, dim3(1), dim3(1), 0, s, draw_buf);
}

extern "C" void launch_value_forward(const float* obs, const void* W1c,
                                     const float* b1c, const void* W2c,
                                     const float* b2c, const void* Wvc,
                                     const float* bvc, float* value_out,
                                     const float* nmean, const float* nvar,
                                     int B, int OBS, int HID, void* stream) {
  hipStream_t s = (hipStream_t)stream;
  dim3 grid((B + 15) / 16), block(256);
  if (HID == 512) {
    hipLaunchKernelGGL(value_forward_kernel<512>, grid, block, 0, s, obs,
                       (const bf16_t*)W1c, b1c, (const bf16_t*)W2c, b2c,
                       (const bf16_t*)Wvc, bvc, value_out, nmean, nvar, B,
                       OBS);
  } else if (HID == 256) {
    hipLaunchKernelGGL(value_forward_kernel<256>, grid, block, 0, s, obs,
                       (const bf16_t*)W1c, b1c, (const bf16_t*)W2c, b2c,
                       (const bf16_t*)Wvc, bvc, value_out, nmean, nvar, B,
                       OBS);
  } else {
    hipLaunchKernelGGL(value_forward_kernel<128>, grid, block, 0, s, obs,
                       (const bf16_t*)W1c, b1c, (const bf16_t*)W2c, b2c,
                       (const bf16_t*)Wvc, bvc, value_out, nmean, nvar, B,
                       OBS);
  }
}

extern "C" void launch_silu_fwd(const void* z, void* h, long n, void* stream) {
  int threads = 256;
  long want = (n / 8 + threads - 1) / threads;
  int blocks = (int)(want < 4096 ? (want > 0 ? want : 1) : 4096);
  hipLaunchKernelGGL(silu_fwd_kernel, dim3(blocks), dim3(threads), 0,
                     (hipStream_t)stream, (const bf16_t*)z, (bf16_t*)h, n);
}

extern "C" void launch_silu_bwd(const void* dh, const void* z, void* dz,
                                long n, void* stream) {
  int threads = 256;
  long want = (n / 8 + threads - 1) / threads;
  int blocks = (int)(want < 4096 ? (want > 0 ? want : 1) : 4096);
  hipLaunchKernelGGL(silu_bwd_kernel, dim3(blocks), dim3(threads), 0,
                     (hipStream_t)stream, (const bf16_t*)dh, (const bf16_t*)z,
                     (bf16_t*)dz, n);
}

extern "C" void launch_ppo_gather(const long* idx, int mb_size,
                                  const float* obs, int OBS, int OBS_PAD,
                                  const float* action, int ACT,
                                  const float* logp, const float* value,
                                  const float* adv, const float* targets,
                                  void* obs_out, float* action_out,
                                  float* logp_out, float* value_out,
                                  float* adv_out, float* targets_out,
                                  const float* nmean, const float* nvar,
                                  void* stream) {
  int threads = 256;
  int rows_per_block = threads / 64;
  int blocks = (mb_size + rows_per_block - 1) / rows_per_block;
  hipLaunchKernelGGL(ppo_gather_kernel, dim3(blocks), dim3(threads), 0,
                     (hipStream_t)stream, idx, mb_size, obs, OBS, OBS_PAD,
                     action, ACT, logp, value, adv, targets, (bf16_t*)obs_out,
                     action_out, logp_out, value_out, adv_out, targets_out,
                     nmean, nvar);
}

extern "C" void launch_ppo_head_loss(
    const void* heads, const void* v_in, const float* action,
    const float* old_logp, const float* old_value, const float* adv,
    const float* targets, void* dhead, void* dv_out, void* dv16_out,
    float* metrics, int B,
    int ACT, float clip_eps, float ent_coef, float vf_coef, float min_scale,
    float aff_scale, float aff_shift, float log_aff_scale, uint64_t seed,
    unsigned int* draw_buf, unsigned int draw_offset, int do_bump,
    void* stream) {
  hipStream_t s = (hipStream_t)stream;
  int threads = 256;
  int blocks = (B + threads - 1) / threads;
  hipLaunchKernelGGL(ppo_head_loss_kernel, dim3(blocks), dim3(threads), 0, s,
                     (const bf16_t*)heads, (const bf16_t*)v_in, action,
                     old_logp, old_value, adv, targets, (bf16_t*)dhead,
                     (bf16_t*)dv_out, (bf16_t*)dv16_out, metrics, B, ACT,
                     clip_eps, ent_coef,
                     vf_coef, min_scale, aff_scale, aff_shift, log_aff_scale,
                     1.0f / (float)B, seed, draw_buf, draw_offset);
  if (draw_buf && do_bump)
    hipLaunchKernelGGL(bump_u32_kernel2, dim3(1), dim3(1), 0, s, draw_buf);
}

// ===================================================================
// Discrete (categorical) fused PPO path — BASELINE config #1 class
// (discrete CartPole / MinAtar heads; reference heads.py:30-41
// CategoricalHead). Same torso machinery as the tanh-normal kernels;
// the epilogues differ: Gumbel-max sampling + log-softmax in the
// rollout, exact categorical entropy + softmax-jacobian head backward
// in the loss kernel. ACT <= 16 (one MFMA N-tile of logits).
// ===================================================================

template <int HID>
__launch_bounds__(256, 2) __global__ void policy_value_step_disc_kernel(
    const float* __restrict__ obs,        // [B, OBS]
    const bf16_t* __restrict__ W1a, const float* __restrict__ b1a,
    const bf16_t* __restrict__ W2a, const float* __restrict__ b2a,
    const bf16_t* __restrict__ Wha, const float* __restrict__ bha,
    const bf16_t* __restrict__ W1c, const float* __restrict__ b1c,
    const bf16_t* __restrict__ W2c, const float* __restrict__ b2c,
    const bf16_t* __restrict__ Wvc, const float* __restrict__ bvc,
    float* __restrict__ obs_mirror,       // [B, OBS] or null
    long* __restrict__ action_out,        // [B] int64
    float* __restrict__ logp_out,         // [B]
    float* __restrict__ value_out,        // [B]
    const float* __restrict__ nmean, const float* __restrict__ nvar,
    int B, int OBS, int ACT, int greedy,
    uint64_t seed, const unsigned int* __restrict__ draw_buf,
    unsigned int draw_offset) {
  __shared__ MlpLds<HID> lds;
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int rbase = blockIdx.x * 16;
  if (rbase >= B) return;
  const uint32_t draw = (draw_buf ? *draw_buf : 0u) + draw_offset;
  const int K1P = (OBS + 31) & ~31;

  bf16_t* O = &lds.O[0][0];
  constexpr int OS = K1P_MAX + OPAD;
  constexpr int HS = HID + HPAD;
  bf16_t* Ha0 = &lds.H[0][0][0];
  bf16_t* Ha1 = &lds.H[1][0][0];
  bf16_t* Hc0 = &lds.H[2][0][0];
  bf16_t* Hc1 = &lds.H[3][0][0];

  stage_obs(obs, rbase, OBS, K1P, O, OS, nmean, nvar, obs_mirror,
            threadIdx.x, 256, B);
  __syncthreads();

  wg_layer<HID, true>(O, OS, W1a, b1a, K1P, Ha0, HS, lane, wid);
  wg_layer<HID, true>(O, OS, W1c, b1c, K1P, Hc0, HS, lane, wid);
  __syncthreads();
  wg_layer<HID, true>(Ha0, HS, W2a, b2a, HID, Ha1, HS, lane, wid);
  wg_layer<HID, true>(Hc0, HS, W2c, b2c, HID, Hc1, HS, lane, wid);
  __syncthreads();

  if (wid == 1) {
    wave_value_head<HID>(Hc1, HS, Wvc, bvc ? *bvc : 0.0f, rbase, value_out,
                         lane, B);
  }

  // ---- actor head: logits tile [16 rows x 16 cols] -> per-row
  // log-softmax + Gumbel-max sample (wave 0)
  if (wid == 0) {
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    const int arow = lane & 15, ak0 = (lane >> 4) * 8;
    for (int ks = 0; ks < HID / 32; ++ks) {
      const bf16x8 a =
          *reinterpret_cast<const bf16x8*>(Ha1 + arow * HS + ks * 32 + ak0);
      bf16x8 b = load_w_frag<HID>(Wha, HID, 0, ks, lane);
      acc = MFMA_BF16_16x16x32(a, b, acc, 0, 0, 0);
    }
    const int col = lane & 15;
    const int g = lane >> 4;
    const bool valid = col < ACT;
    float bh = bha[col];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int grow = rbase + g * 4 + r;
      float logit = valid ? acc[r] + bh : -3.0e38f;
      // log-sum-exp over the 16-lane column group
      float m = logit;
#pragma unroll
      for (int off = 1; off < 16; off <<= 1) m = fmaxf(m, __shfl_xor(m, off));
      float e = valid ? __expf(logit - m) : 0.0f;
      float sum = e;
#pragma unroll
      for (int off = 1; off < 16; off <<= 1) sum += __shfl_xor(sum, off);
      float lse = m + __logf(sum);
      // Gumbel-max sample (one philox block serves 4 rows per col)
      float score = logit;
      if (!greedy && valid) {
        int rowblk = (rbase + g * 4) >> 2;
        Rng4 u = philox_uniform4(seed, 2u, (uint32_t)(rowblk * 16 + col), draw);
        float uu = (r == 0) ? u.a : (r == 1) ? u.b : (r == 2) ? u.c : u.d;
        uu = fmaxf(uu, 1e-12f);
        score = logit - __logf(-__logf(uu));
      }
      if (!valid) score = -3.0e38f;
      // argmax over the group with index
      float s = score;
      int arg = col;
#pragma unroll
      for (int off = 1; off < 16; off <<= 1) {
        float so = __shfl_xor(s, off);
        int ao = __shfl_xor(arg, off);
        if (so > s || (so == s && ao < arg)) { s = so; arg = ao; }
      }
      float logit_a = __shfl(logit, (lane & 48) | arg, 64);
      if (grow < B && col == 0) {
        action_out[grow] = (long)arg;
        logp_out[grow] = logit_a - lse;
      }
    }
  }
}

// Per-row categorical PPO head loss + analytic backward (see the
// tanh-normal ppo_head_loss_kernel above for the shared clip/value math;
// gradients here go through the softmax jacobian:
// d logp_a / d logits_j = delta_aj - p_j;
// dH / d logits_j = -p_j (logp_j + H)).
extern "C" __global__ void ppo_head_loss_disc_kernel(
    const bf16_t* __restrict__ heads,    // [B, 16] logits (cols >= ACT pad)
    const bf16_t* __restrict__ v_in,     // [B]
    const long* __restrict__ action,     // [B] int64
    const float* __restrict__ old_logp,  // [B]
    const float* __restrict__ old_value, // [B]
    const float* __restrict__ adv,       // [B]
    const float* __restrict__ targets,   // [B]
    bf16_t* __restrict__ dhead,          // [B, 16]
    bf16_t* __restrict__ dv_out,         // [B]
    bf16_t* __restrict__ dv16_out,       // [B,16] col 0 or null
    float* __restrict__ metrics,         // [3] or null
    int B, int ACT, float clip_eps, float ent_coef, float vf_coef,
    float inv_B) {
  const int row = blockIdx.x * blockDim.x + threadIdx.x;
  const bool active = row < B;

  float l[16], p[16];
  float lse = 0.0f, Hent = 0.0f, logp_new = 0.0f;
  int a = 0;
  if (active) {
    const bf16x8* h8 = reinterpret_cast<const bf16x8*>(heads + (long)row * 16);
    bf16x8 h0 = h8[0], h1 = h8[1];
#pragma unroll
    for (int j = 0; j < 8; ++j) { l[j] = bf2f(h0[j]); l[8 + j] = bf2f(h1[j]); }
    a = (int)action[row];
    float m = -3.0e38f;
    for (int j = 0; j < ACT; ++j) m = fmaxf(m, l[j]);
    float sum = 0.0f;
    for (int j = 0; j < ACT; ++j) sum += __expf(l[j] - m);
    lse = m + __logf(sum);
    for (int j = 0; j < 16; ++j) p[j] = (j < ACT) ? __expf(l[j] - lse) : 0.0f;
    for (int j = 0; j < ACT; ++j) Hent -= p[j] * (l[j] - lse);
    logp_new = l[a] - lse;
  }

  // ---- PPO clip loss (identical to the continuous kernel)
  float A = active ? adv[row] : 0.0f;
  float ratio = __expf(logp_new - (active ? old_logp[row] : 0.0f));
  float r_clip = fmaxf(1.0f - clip_eps, fminf(1.0f + clip_eps, ratio));
  float l1 = ratio * A, l2 = r_clip * A;
  float a_loss = -fminf(l1, l2);
  float dl_dlogp;
  if (l1 <= l2) {
    dl_dlogp = -ratio * A;
  } else {
    dl_dlogp = (ratio > 1.0f - clip_eps && ratio < 1.0f + clip_eps)
                   ? -ratio * A
                   : 0.0f;
  }
  dl_dlogp *= inv_B;

  // ---- clipped value loss (identical)
  float v_pred = active ? bf2f(v_in[row]) : 0.0f;
  float ov = active ? old_value[row] : 0.0f;
  float tg = active ? targets[row] : 0.0f;
  float v_clip = ov + fmaxf(-clip_eps, fminf(clip_eps, v_pred - ov));
  float e1 = (v_pred - tg), e2 = (v_clip - tg);
  float sq1 = e1 * e1, sq2 = e2 * e2;
  float v_loss = 0.5f * fmaxf(sq1, sq2);
  float dv;
  if (sq1 >= sq2) {
    dv = e1;
  } else {
    dv = (fabsf(v_pred - ov) < clip_eps) ? e2 : 0.0f;
  }
  dv *= vf_coef * inv_B;

  // ---- analytic logits gradient
  if (active) {
    const float ce = ent_coef * inv_B;
    bf16x8 d0, d1;
#pragma unroll
    for (int j = 0; j < 16; ++j) {
      float g = 0.0f;
      if (j < ACT) {
        float delta = (j == a) ? 1.0f : 0.0f;
        g = dl_dlogp * (delta - p[j]);
        g += ce * p[j] * ((l[j] - lse) + Hent);
      }
      if (j < 8) d0[j] = f2bf(g); else d1[j - 8] = f2bf(g);
    }
    bf16x8* out8 = reinterpret_cast<bf16x8*>(dhead + (long)row * 16);
    out8[0] = d0;
    out8[1] = d1;
    dv_out[row] = f2bf(dv);
    if (dv16_out) dv16_out[(long)row * 16] = f2bf(dv);
  }

  if (metrics) {
    float m0 = active ? a_loss * inv_B : 0.0f;
    float m1 = active ? v_loss * inv_B : 0.0f;
    float m2 = active ? Hent * inv_B : 0.0f;
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      m0 += __shfl_down(m0, off);
      m1 += __shfl_down(m1, off);
      m2 += __shfl_down(m2, off);
    }
    if ((threadIdx.x & 63) == 0) {
      atomicAdd(&metrics[0], m0);
      atomicAdd(&metrics[1], m1);
      atomicAdd(&metrics[2], m2);
    }
  }
}

// Minibatch gather for the discrete path: int64 actions, no per-dim loop.
extern "C" __global__ void ppo_gather_disc_kernel(
    const long* __restrict__ idx, int mb_size,
    const float* __restrict__ obs, int OBS, int OBS_PAD,
    const long* __restrict__ action,
    const float* __restrict__ logp, const float* __restrict__ value,
    const float* __restrict__ adv, const float* __restrict__ targets,
    bf16_t* __restrict__ obs_out, long* __restrict__ action_out,
    float* __restrict__ logp_out, float* __restrict__ value_out,
    float* __restrict__ adv_out, float* __restrict__ targets_out,
    const float* __restrict__ nmean, const float* __restrict__ nvar) {
  int row = (blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  int lane = threadIdx.x & 63;
  if (row >= mb_size) return;
  long src = idx[row];
  for (int k = lane; k < OBS_PAD; k += 64) {
    float v = 0.0f;
    if (k < OBS) {
      v = obs[src * OBS + k];
      if (nmean) {
        float sd = sqrtf(fmaxf(nvar[k], 1e-6f));
        v = fmaxf(-10.0f, fminf(10.0f, (v - nmean[k]) / sd));
      }
    }
    obs_out[(long)row * OBS_PAD + k] = f2bf(v);
  }
  if (lane == 0) {
    action_out[row] = action[src];
    logp_out[row] = logp[src];
    value_out[row] = value[src];
    adv_out[row] = adv[src];
    targets_out[row] = targets[src];
  }
}

extern "C" void launch_policy_value_step_disc(
    const float* obs, const void* W1a, const float* b1a, const void* W2a,
    const float* b2a, const void* Wha, const float* bha, const void* W1c,
    const float* b1c, const void* W2c, const float* b2c, const void* Wvc,
    const float* bvc, float* obs_mirror, long* action_out, float* logp_out,
    float* value_out, const float* nmean, const float* nvar, int B, int OBS,
    int ACT, int HID, int greedy, uint64_t seed, unsigned int* draw_buf,
    unsigned int draw_offset, int do_bump, void* stream) {
  hipStream_t s = (hipStream_t)stream;
  dim3 grid((B + 15) / 16), block(256);
  if (HID == 512) {
    hipLaunchKernelGGL(policy_value_step_disc_kernel<512>, grid, block, 0, s,
                       obs, (const bf16_t*)W1a, b1a, (const bf16_t*)W2a, b2a,
                       (const bf16_t*)Wha, bha, (const bf16_t*)W1c, b1c,
                       (const bf16_t*)W2c, b2c, (const bf16_t*)Wvc, bvc,
                       obs_mirror, action_out, logp_out, value_out, nmean,
                       nvar, B, OBS, ACT, greedy, seed, draw_buf, draw_offset);
  } else if (HID == 256) {
    hipLaunchKernelGGL(policy_value_step_disc_kernel<256>, grid, block, 0, s,
                       obs, (const bf16_t*)W1a, b1a, (const bf16_t*)W2a, b2a,
                       (const bf16_t*)Wha, bha, (const bf16_t*)W1c, b1c,
                       (const bf16_t*)W2c, b2c, (const bf16_t*)Wvc, bvc,
                       obs_mirror, action_out, logp_out, value_out, nmean,
                       nvar, B, OBS, ACT, greedy, seed, draw_buf, draw_offset);
  } else {
    hipLaunchKernelGGL(policy_value_step_disc_kernel<128>, grid, block, 0, s,
                       obs, (const bf16_t*)W1a, b1a, (const bf16_t*)W2a, b2a,
                       (const bf16_t*)Wha, bha, (const bf16_t*)W1c, b1c,
                       (const bf16_t*)W2c, b2c, (const bf16_t*)Wvc, bvc,
                       obs_mirror, action_out, logp_out, value_out, nmean,
                       nvar, B, OBS, ACT, greedy, seed, draw_buf, draw_offset);
  }
  if (draw_buf && do_bump)
    hipLaunchKernelGGL(bump_u32_kernel2, dim3(1), dim3(1), 0, s, draw_buf);
}

extern "C" void launch_ppo_head_loss_disc(
    const void* heads, const void* v_in, const long* action,
    const float* old_logp, const float* old_value, const float* adv,
    const float* targets, void* dhead, void* dv_out, void* dv16_out,
    float* metrics, int B, int ACT, float clip_eps, float ent_coef,
    float vf_coef, void* stream) {
  hipStream_t s = (hipStream_t)stream;
  int threads = 256;
  int blocks = (B + threads - 1) / threads;
  hipLaunchKernelGGL(ppo_head_loss_disc_kernel, dim3(blocks), dim3(threads),
                     0, s, (const bf16_t*)heads, (const bf16_t*)v_in, action,
                     old_logp, old_value, adv, targets, (bf16_t*)dhead,
                     (bf16_t*)dv_out, (bf16_t*)dv16_out, metrics, B, ACT,
                     clip_eps, ent_coef, vf_coef, 1.0f / (float)B);
}

extern "C" void launch_ppo_gather_disc(
    const long* idx, int mb_size, const float* obs, int OBS, int OBS_PAD,
    const long* action, const float* logp, const float* value,
    const float* adv, const float* targets, void* obs_out, long* action_out,
    float* logp_out, float* value_out, float* adv_out, float* targets_out,
    const float* nmean, const float* nvar, void* stream) {
  int threads = 256;
  int rows_per_block = threads / 64;
  int blocks = (mb_size + rows_per_block - 1) / rows_per_block;
  hipLaunchKernelGGL(ppo_gather_disc_kernel, dim3(blocks), dim3(threads), 0,
                     (hipStream_t)stream, idx, mb_size, obs, OBS, OBS_PAD,
                     action, logp, value, adv, targets, (bf16_t*)obs_out,
                     action_out, logp_out, value_out, adv_out, targets_out,
                     nmean, nvar);
}
