// K13: fused done-masked RNN sequence scan.
//
// The reference's ScannedRNN (stoix/networks/base.py:124-159) is an
// nn.scan of a GRU/LSTM cell with per-step hidden reset on done. The
// recurrence is sequential in T but per-ROW independent, so the MI355X
// mapping is: one 16-row tile per workgroup that owns its rows for the
// WHOLE T loop — hidden state lives in LDS (fp32 master + bf16 mirror for
// MFMA) and never leaves the CU between steps; the input-side projection
// Xp = x @ W_ih^T + b_ih for ALL T is hoisted into one big hipBLASLt GEMM
// by the caller (the classic cuDNN trick), so the kernel does only the
// recurrent half: per step one [16,H]x[H,G*H] MFMA GEMM + gate math.
//
// Gate math matches nn.GRUCell / nn.LSTMCell exactly:
//   GRU  (gates r|z|n):  r = s(xr+hr)  z = s(xz+hz)
//                        n = tanh(xn + r*(h Whn + bhn));  h' = (1-z)n + z h
//   LSTM (gates i|f|g|o): c' = s(f)*c + s(i)*tanh(g);  h' = s(o)*tanh(c')
// The keep-mask (reset-before-consume) is applied to the fp32 master and
// the bf16 mirror at the top of each step.
//
// This is the no-grad path (rollout acting, evaluation, R2D2 burn-in —
// which the reference runs under stop_gradient, rec_r2d2.py:300-317). The
// training scan stays on autograd with the same hoisted-Xp structure
// (networks/base.py ScannedRNN.forward).
#include "common.h"
#include <hip/hip_bf16.h>

typedef __bf16 bf16_t;
typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

#define MFMA_BF16_16x16x32 __builtin_amdgcn_mfma_f32_16x16x32_bf16

DEV_INLINE float bf2fr(bf16_t x) { return (float)x; }
DEV_INLINE bf16_t f2bfr(float x) { return (bf16_t)x; }
DEV_INLINE float sigm(float x) { return 1.0f / (1.0f + __expf(-x)); }
DEV_INLINE float tanh_r(float x) {
  return 1.0f - 2.0f / (__expf(2.0f * x) + 1.0f);
}

// B fragment loader: W row-major [G*H, H] bf16; tile row base `nrow`.
DEV_INLINE bf16x8 rnn_w_frag(const bf16_t* __restrict__ W, int K, int nrow,
                             int ks, int lane) {
  int n = nrow + (lane & 15);
  int k0 = ks * 32 + (lane >> 4) * 8;
  return *reinterpret_cast<const bf16x8*>(W + (long)n * K + k0);
}

constexpr int RPAD = 8;  // bf16 mirror column pad (conflict-free b128 reads)

template <int H, int GATES>
struct RnnLds {
  float hf[16][H];
  bf16_t hb[16][H + RPAD];
  float cf[16][GATES == 4 ? H : 1];  // LSTM cell state
};

template <int H, bool LSTM>
__launch_bounds__(256, 2) __global__ void rnn_scan_kernel(
    const float* __restrict__ Xp,      // [T, B, G*H] input projections
    const bf16_t* __restrict__ Whh,    // [G*H, H] bf16
    const float* __restrict__ bhh,     // [G*H]
    const unsigned char* __restrict__ resets,  // [T, B]
    const float* __restrict__ h0,      // [B, H]
    const float* __restrict__ c0,      // [B, H] (LSTM) or null
    float* __restrict__ Hout,          // [T, B, H]
    float* __restrict__ hT,            // [B, H]
    float* __restrict__ cT,            // [B, H] (LSTM) or null
    int T, int B) {
  constexpr int G = LSTM ? 4 : 3;
  __shared__ RnnLds<H, G> lds;
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int rbase = blockIdx.x * 16;
  if (rbase >= B) return;
  constexpr int STRIP = H / 4;       // columns per wave
  constexpr int NTW = STRIP / 16;    // 16-col tiles per gate per wave
  const int strip0 = wid * STRIP;

  // stage h0 (and c0)
  for (int i = threadIdx.x; i < 16 * H; i += 256) {
    int r = i / H, k = i - r * H;
    float v = (rbase + r < B) ? h0[(long)(rbase + r) * H + k] : 0.0f;
    lds.hf[r][k] = v;
    lds.hb[r][k] = f2bfr(v);
    if (LSTM) {
      float cv = (rbase + r < B && c0) ? c0[(long)(rbase + r) * H + k] : 0.0f;
      lds.cf[r][k] = cv;
    }
  }
  __syncthreads();

  const int arow = lane & 15;
  const int ak0 = (lane >> 4) * 8;
  const int col = lane & 15;
  const int grp = lane >> 4;

  for (int t = 0; t < T; ++t) {
    // ---- reset-before-consume mask
    for (int i = threadIdx.x; i < 16 * H; i += 256) {
      int r = i / H, k = i - r * H;
      unsigned char rs =
          (rbase + r < B) ? resets[(long)t * B + rbase + r] : 0;
      if (rs) {
        lds.hf[r][k] = 0.0f;
        lds.hb[r][k] = f2bfr(0.0f);
        if (LSTM) lds.cf[r][k] = 0.0f;
      }
    }
    __syncthreads();

    // ---- recurrent GEMM: gates for this wave's column strip
    f32x4 acc[4][NTW];  // [gate][tile]
#pragma unroll
    for (int g = 0; g < G; ++g)
#pragma unroll
      for (int n = 0; n < NTW; ++n) acc[g][n] = {0.f, 0.f, 0.f, 0.f};
    for (int ks = 0; ks < H / 32; ++ks) {
      const bf16x8 a = *reinterpret_cast<const bf16x8*>(
          &lds.hb[arow][0] + ks * 32 + ak0);
#pragma unroll
      for (int g = 0; g < G; ++g) {
#pragma unroll
        for (int n = 0; n < NTW; ++n) {
          bf16x8 b = rnn_w_frag(Whh, H, g * H + strip0 + n * 16, ks, lane);
          acc[g][n] = MFMA_BF16_16x16x32(a, b, acc[g][n], 0, 0, 0);
        }
      }
    }
    // ---- read own h_old (and c_old) BEFORE the write barrier
    float hold[4][NTW], cold[4][NTW];
#pragma unroll
    for (int n = 0; n < NTW; ++n) {
      int cg = strip0 + n * 16 + col;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        hold[r][n] = lds.hf[grp * 4 + r][cg];
        if (LSTM) cold[r][n] = lds.cf[grp * 4 + r][cg];
      }
    }
    __syncthreads();

    // ---- gate math + state update + output write
#pragma unroll
    for (int n = 0; n < NTW; ++n) {
      int cg = strip0 + n * 16 + col;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = grp * 4 + r;
        int grow = rbase + row;
        if (grow >= B) continue;
        const long xbase = ((long)t * B + grow) * (G * H);
        float hnew;
        if (!LSTM) {
          float xr = Xp[xbase + cg];
          float xz = Xp[xbase + H + cg];
          float xn = Xp[xbase + 2 * H + cg];
          float hr = acc[0][n][r] + bhh[cg];
          float hz = acc[1][n][r] + bhh[H + cg];
          float hn = acc[2][n][r] + bhh[2 * H + cg];
          float rg = sigm(xr + hr);
          float zg = sigm(xz + hz);
          float ng = tanh_r(xn + rg * hn);
          hnew = (1.0f - zg) * ng + zg * hold[r][n];
        } else {
          float xi = Xp[xbase + cg];
          float xf = Xp[xbase + H + cg];
          float xg = Xp[xbase + 2 * H + cg];
          float xo = Xp[xbase + 3 * H + cg];
          float ig = sigm(xi + acc[0][n][r] + bhh[cg]);
          float fg = sigm(xf + acc[1][n][r] + bhh[H + cg]);
          float gg = tanh_r(xg + acc[2][n][r] + bhh[2 * H + cg]);
          float og = sigm(xo + acc[3][n][r] + bhh[3 * H + cg]);
          float cnew = fg * cold[r][n] + ig * gg;
          lds.cf[row][cg] = cnew;
          hnew = og * tanh_r(cnew);
        }
        lds.hf[row][cg] = hnew;
        lds.hb[row][cg] = f2bfr(hnew);
        Hout[((long)t * B + grow) * H + cg] = hnew;
      }
    }
    __syncthreads();
  }

  // ---- final state
  for (int i = threadIdx.x; i < 16 * H; i += 256) {
    int r = i / H, k = i - r * H;
    if (rbase + r < B) {
      hT[(long)(rbase + r) * H + k] = lds.hf[r][k];
      if (LSTM && cT) cT[(long)(rbase + r) * H + k] = lds.cf[r][k];
    }
  }
}

// --------------------------------------------------------- host launchers

extern "C" void launch_rnn_scan(const float* Xp, const void* Whh,
                                const float* bhh,
                                const unsigned char* resets, const float* h0,
                                const float* c0, float* Hout, float* hT,
                                float* cT, int T, int B, int H, int lstm,
                                void* stream) {
  hipStream_t s = (hipStream_t)stream;
  dim3 grid((B + 15) / 16), block(256);
  if (H == 256) {
    if (lstm)
      hipLaunchKernelGGL((rnn_scan_kernel<256, true>), grid, block, 0, s, Xp,
                         (const bf16_t*)Whh, bhh, resets, h0, c0, Hout, hT,
                         cT, T, B);
    else
      hipLaunchKernelGGL((rnn_scan_kernel<256, false>), grid, block, 0, s,
                         Xp, (const bf16_t*)Whh, bhh, resets, h0, c0, Hout,
                         hT, cT, T, B);
  } else {
    if (lstm)
      hipLaunchKernelGGL((rnn_scan_kernel<128, true>), grid, block, 0, s, Xp,
                         (const bf16_t*)Whh, bhh, resets, h0, c0, Hout, hT,
                         cT, T, B);
    else
      hipLaunchKernelGGL((rnn_scan_kernel<128, false>), grid, block, 0, s,
                         Xp, (const bf16_t*)Whh, bhh, resets, h0, c0, Hout,
                         hT, cT, T, B);
  }
}
