// Split-K weight-gradient kernels (the K6/K7 backward of SURVEY.md §2.9).
//
// dW[N,K] = dZ^T[N,S] @ X[S,K]  and  db[N] = colsum(dZ),  S = minibatch
// (32768 for the flagship config). hipBLASLt's best NT kernel for this
// K-huge/MN-tiny shape runs at ~56 TFLOP/s (tunableop_gfx950.csv:
// nt_256_256_32768 = 77 us); these kernels split S over 16 wave-slices,
// each wave MFMA-accumulating its slice into a private fp32 slab, and one
// slab_reduce kernel sums the 16 slabs straight into the flat bf16 grad
// buffer (zeroing the slab for the next minibatch in the same pass). The
// bias column-sum rides along for free from the A-operand fragments.
//
// Geometry per wgrad launch: grid (N/16) x (K/(16*KPG)) x 4, block 256
// (4 waves). wave-slice = blockIdx.z*4 + wid in [0,16); each wave stages
// 32-deep s-tiles of dZ and X into LDS *transposed* (scatter b16 writes,
// contiguous ds_read_b128 fragment reads) and issues KPG
// mfma_f32_16x16x32_bf16 per s-step.
#include "common.h"
#include <hip/hip_bf16.h>

typedef __bf16 bf16_t;
typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef __bf16 bf16x4 __attribute__((ext_vector_type(4)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

#define MFMA_BF16_16x16x32 __builtin_amdgcn_mfma_f32_16x16x32_bf16

#define WG_SLICES 16  // 4 z-blocks x 4 waves

// Per-wave staging, ROW-MAJOR [32 s][16 x] subtiles: global loads and the
// LDS writes are both 16B-contiguous (no transposed scatter), and the MFMA
// fragments are read with the gfx950 hardware transpose-read
// ds_read_tr16_b64 (semantics decoded empirically, tools/tr16_analyze.py:
// within each quad of lanes, lane 4q+i element j = 16-bit element i at
// lane 4q+j's address -> a 4x4 transpose per quad). For an operand tile
// [32 s][16 x], per-lane address row (l>>4)*8+(l&3)+s_half*4, column
// 4*((l>>2)&3) yields exactly the MFMA fragment x = l&15, s = (l>>4)*8+j.
struct WgradLds {
  bf16_t dZt[4][32][16];
  bf16_t Xt[4][8][32][16];
};

typedef __attribute__((address_space(3))) bf16x4 lds_bf16x4;

DEV_INLINE bf16x4 tr_read(const bf16_t* base) {
  return __builtin_amdgcn_ds_read_tr16_b64_v4bf16((lds_bf16x4*)base);
}

// Read one [32 s] x [16 x] row-major LDS tile as an MFMA A/B fragment
// (lane l -> x = l&15, s = (l>>4)*8 + j, j = 0..7) via two tr reads.
DEV_INLINE bf16x8 frag_from_tile(const bf16_t (*tile)[16], int lane) {
  int row = (lane >> 4) * 8 + (lane & 3);
  int col = 4 * ((lane >> 2) & 3);
  bf16x4 lo = tr_read(&tile[row][col]);
  bf16x4 hi = tr_read(&tile[row + 4][col]);
  bf16x8 out;
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    out[j] = lo[j];
    out[4 + j] = hi[j];
  }
  return out;
}

template <int KPG>
__launch_bounds__(256, 2) __global__ void wgrad_kernel(
    const bf16_t* __restrict__ dZ,  // [S, N_STRIDE]
    const bf16_t* __restrict__ X,   // [S, K]
    float* __restrict__ slab,       // [WG_SLICES, slab_stride] fp32
    long dW_off,                    // element offset of dW[N,K] in a slab
    long db_off,                    // element offset of db[N] (or -1)
    long slab_stride, int S, int N_STRIDE, int K, int N_VALID) {
  __shared__ WgradLds lds;
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int nt = blockIdx.x;
  const int k0 = blockIdx.y * KPG * 16;
  const int slice = blockIdx.z * 4 + wid;
  const int s_per = S / WG_SLICES;
  const int s_begin = slice * s_per;
  const int s_end = s_begin + s_per;
  const int n0 = nt * 16;

  bf16_t(*dZt)[16] = lds.dZt[wid];
  bf16_t(*Xt)[32][16] = lds.Xt[wid];

  f32x4 acc[KPG];
#pragma unroll
  for (int t = 0; t < KPG; ++t) acc[t] = {0.f, 0.f, 0.f, 0.f};
  float db_acc = 0.0f;

  for (int s0 = s_begin; s0 < s_end; s0 += 32) {
    // ---- stage dZ tile [32 s][16 n], row-major (1 x 16B load+write/lane)
    {
      int srow = lane >> 1, nh = (lane & 1) * 8;
      *reinterpret_cast<bf16x8*>(&dZt[srow][nh]) =
          *reinterpret_cast<const bf16x8*>(dZ + (long)(s0 + srow) * N_STRIDE +
                                           n0 + nh);
    }
    // ---- stage X subtiles [KPG][32 s][16 k], row-major
#pragma unroll
    for (int q = 0; q < KPG; ++q) {
      int flat = q * 64 + lane;
      int srow = flat / (KPG * 2);
      int hc = flat % (KPG * 2);
      *reinterpret_cast<bf16x8*>(&Xt[hc >> 1][srow][(hc & 1) * 8]) =
          *reinterpret_cast<const bf16x8*>(X + (long)(s0 + srow) * K + k0 +
                                           hc * 8);
    }
    // within-wave LDS write->read ordering is compiler-tracked (lgkmcnt)
    const bf16x8 a = frag_from_tile(dZt, lane);
#pragma unroll
    for (int t = 0; t < KPG; ++t) {
      const bf16x8 b = frag_from_tile(Xt[t], lane);
      acc[t] = MFMA_BF16_16x16x32(a, b, acc[t], 0, 0, 0);
    }
    if (db_off >= 0 && blockIdx.y == 0) {
#pragma unroll
      for (int j = 0; j < 8; ++j) db_acc += (float)a[j];
    }
  }

  float* out = slab + (long)slice * slab_stride;
  const int col = lane & 15;
  const int g = lane >> 4;
#pragma unroll
  for (int t = 0; t < KPG; ++t) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int n = g * 4 + r;
      if (n < N_VALID)
        out[dW_off + (long)(n0 + n) * K + k0 + t * 16 + col] = acc[t][r];
    }
  }
  if (db_off >= 0 && blockIdx.y == 0) {
    db_acc += __shfl_xor(db_acc, 16);
    db_acc += __shfl_xor(db_acc, 32);
    if ((lane >> 4) == 0 && (lane & 15) < N_VALID)
      out[db_off + n0 + (lane & 15)] = db_acc;
  }
}

// Sum the WG_SLICES slabs into the flat bf16 grad buffer and zero them for
// the next minibatch. One launch covers a whole chain's gradient.
extern "C" __global__ void slab_reduce_kernel(float* __restrict__ slab,
                                              __bf16* __restrict__ grad16,
                                              long slab_stride, long n,
                                              float* __restrict__ sqnorm,
                                              long* __restrict__ step_t) {
  // fused Adam prologue (zero the norm accumulator, bump the step counter)
  // rides along: this kernel always runs right before the chain's Adam
  if (sqnorm && blockIdx.x == 0 && threadIdx.x == 0) {
    *sqnorm = 0.0f;
    *step_t += 1;
  }
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    float s = 0.0f;
#pragma unroll
    for (int z = 0; z < WG_SLICES; ++z) {
      s += slab[(long)z * slab_stride + i];
      slab[(long)z * slab_stride + i] = 0.0f;
    }
    grad16[i] = (__bf16)s;
  }
}

// ------------------------------------------------------------- tr probe
// Empirically maps __builtin_amdgcn_ds_read_tr16_b64_v4bf16: stages 1024
// known bf16 values linearly in LDS, issues the tr read with per-lane base
// = base_mode ? lane-pattern : uniform, dumps each lane's 4 elements.
extern "C" __global__ void tr16_probe_kernel(const bf16_t* __restrict__ in,
                                             float* __restrict__ out,
                                             int base_mode) {
  __shared__ bf16_t l[1024];
  int t = threadIdx.x;
  for (int i = t; i < 1024; i += 64) l[i] = in[i];
  __syncthreads();
  if (t >= 64) return;
  int off = 0;
  if (base_mode == 1) off = (t >> 4) * 64;       // group-strided base
  else if (base_mode == 2) off = (t >> 4) * 128; // doubled group stride
  bf16x4 v = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
      (__attribute__((address_space(3))) bf16x4*)&l[off]);
#pragma unroll
  for (int j = 0; j < 4; ++j) out[t * 4 + j] = (float)v[j];
}

// --------------------------------------------------------- host launchers

extern "C" void launch_wgrad(const void* dZ, const void* X, float* slab,
                             long dW_off, long db_off, long slab_stride,
                             int S, int N_STRIDE, int K, int N_VALID,
                             void* stream) {
  hipStream_t s = (hipStream_t)stream;
  int NT = (N_VALID + 15) / 16;
  // largest KPG whose grid still spans >= 16 workgroups: small-N shapes
  // (head grads, NT=1) would otherwise run on 8 CUs of 256
  int KPG = 1;
  const int cands[3] = {8, 2, 1};
  for (int ci = 0; ci < 3; ++ci) {
    int cand = cands[ci];
    if (K % (16 * cand) == 0 && (cand == 1 || NT * (K / (16 * cand)) >= 16)) {
      KPG = cand;
      break;
    }
  }
  int KTG = K / (16 * KPG);
  dim3 grid(NT, KTG, 4), block(256);
  if (KPG == 8) {
    hipLaunchKernelGGL(wgrad_kernel<8>, grid, block, 0, s, (const bf16_t*)dZ,
                       (const bf16_t*)X, slab, dW_off, db_off, slab_stride, S,
                       N_STRIDE, K, N_VALID);
  } else if (KPG == 2) {
    hipLaunchKernelGGL(wgrad_kernel<2>, grid, block, 0, s, (const bf16_t*)dZ,
                       (const bf16_t*)X, slab, dW_off, db_off, slab_stride, S,
                       N_STRIDE, K, N_VALID);
  } else {
    hipLaunchKernelGGL(wgrad_kernel<1>, grid, block, 0, s, (const bf16_t*)dZ,
                       (const bf16_t*)X, slab, dW_off, db_off, slab_stride, S,
                       N_STRIDE, K, N_VALID);
  }
}

extern "C" void launch_slab_reduce(float* slab, void* grad16,
                                   long slab_stride, long n, float* sqnorm,
                                   long* step_t, void* stream) {
  int threads = 256;
  long want = (n + threads - 1) / threads;
  int blocks = (int)(want < 2048 ? (want > 0 ? want : 1) : 2048);
  hipLaunchKernelGGL(slab_reduce_kernel, dim3(blocks), dim3(threads), 0,
                     (hipStream_t)stream, slab, (__bf16*)grad16, slab_stride,
                     n, sqnorm, step_t);
}

extern "C" void launch_tr16_probe(const void* in, float* out, int base_mode,
                                  void* stream) {
  hipLaunchKernelGGL(tr16_probe_kernel, dim3(1), dim3(64), 0,
                     (hipStream_t)stream, (const bf16_t*)in, out, base_mode);
}
