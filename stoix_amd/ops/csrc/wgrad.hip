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
#include <cstdlib>
#include <hip/hip_bf16.h>

typedef __bf16 bf16_t;
typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef __bf16 bf16x4 __attribute__((ext_vector_type(4)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

#define MFMA_BF16_16x16x32 __builtin_amdgcn_mfma_f32_16x16x32_bf16

#define WG_SLICES 64  // max wave-slices; the launcher picks the
// largest power of two <= 64 that divides S/32 (small minibatches use fewer)
#define SPAD 8        // +8 cols on the 32-wide transposed tiles

// One wave's private staging: dZt [NTB*16 n][32+8 s], Xt [KPG*16 k][32+8 s].
// NTB n-tiles per workgroup cut the X read-amplification (every n-tile
// group re-reads the whole X k-slice; at NTB=1 a 256x256 wgrad re-reads
// the 16 MB X sixteen times = 256 MB -> the kernel is HBM/L2-traffic
// bound). NTB=4, KPG=8: per-wave 4*16*40*2 + 8*16*40*2 = 15.3 KB; x4
// waves = 61 KB.
struct WgradLds {
  bf16_t dZt[4][4 * 16][32 + SPAD];
  bf16_t Xt[4][8 * 16][32 + SPAD];
};

template <int KPG, int NTB>
__launch_bounds__(256, 2) __global__ void wgrad_kernel(
    const bf16_t* __restrict__ dZ,  // [S, N_STRIDE]
    const bf16_t* __restrict__ X,   // [S, K]
    float* __restrict__ slab,       // [WG_SLICES, slab_stride] fp32
    long dW_off,                    // element offset of dW[N,K] in a slab
    long db_off,                    // element offset of db[N] (or -1)
    long slab_stride, int S, int N_STRIDE, int K, int N_VALID,
    int n_slices) {
  __shared__ WgradLds lds;
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int nt = blockIdx.x * NTB;   // NTB 16-row n-tiles per workgroup
  const int k0 = blockIdx.y * KPG * 16;
  const int slice = blockIdx.z * 4 + wid;  // [0, n_slices)
  const int s_per = S / n_slices;
  const int s_begin = slice * s_per;
  const int s_end = s_begin + s_per;
  const int n0 = nt * 16;

  bf16_t(*dZt)[32 + SPAD] = lds.dZt[wid];
  bf16_t(*Xt)[32 + SPAD] = lds.Xt[wid];

  f32x4 acc[NTB][KPG];
#pragma unroll
  for (int u = 0; u < NTB; ++u)
#pragma unroll
    for (int t = 0; t < KPG; ++t) acc[u][t] = {0.f, 0.f, 0.f, 0.f};
  float db_acc[NTB];
#pragma unroll
  for (int u = 0; u < NTB; ++u) db_acc[u] = 0.0f;

  const int arow = lane & 15;
  const int ak0 = (lane >> 4) * 8;

  for (int s0 = s_begin; s0 < s_end; s0 += 32) {
    // ---- stage dZ tiles [32 s][NTB*16 n] -> dZt [NTB*16 n][32 s]
#pragma unroll
    for (int u = 0; u < NTB; ++u) {
      int srow = lane >> 1, nh = (lane & 1) * 8;
      bf16x8 v = *reinterpret_cast<const bf16x8*>(
          dZ + (long)(s0 + srow) * N_STRIDE + n0 + u * 16 + nh);
#pragma unroll
      for (int i = 0; i < 8; ++i) dZt[u * 16 + nh + i][srow] = v[i];
    }
    // ---- stage X tiles [32 s][KPG*16 k] -> Xt [KPG*16 k][32 s].
    // Lane mapping: consecutive lanes take consecutive s-rows of ONE
    // k-chunk, so the 8 transposed b16 scatter-writes per lane hit
    // consecutive banks (srow-consecutive) instead of a 16-way conflict
    // (kh-strided, 160 dwords = bank 0 for every lane). The global loads
    // become 16B row-strided, but successive q iterations re-touch the
    // same 32 rows' cache lines, so L1 serves 7/8 of them.
#pragma unroll
    for (int q = 0; q < KPG; ++q) {
      int flat = q * 64 + lane;
      int srow = flat & 31;
      int kh = (flat >> 5) * 8;
      bf16x8 v = *reinterpret_cast<const bf16x8*>(
          X + (long)(s0 + srow) * K + k0 + kh);
#pragma unroll
      for (int i = 0; i < 8; ++i) Xt[kh + i][srow] = v[i];
    }
    // within-wave LDS write->read ordering is compiler-tracked (lgkmcnt);
    // no cross-wave sharing, so no barrier.
    // ---- fragments + MFMA (B fragments shared by the NTB A tiles)
#pragma unroll
    for (int t = 0; t < KPG; ++t) {
      const bf16x8 b =
          *reinterpret_cast<const bf16x8*>(&Xt[t * 16 + arow][ak0]);
#pragma unroll
      for (int u = 0; u < NTB; ++u) {
        const bf16x8 a =
            *reinterpret_cast<const bf16x8*>(&dZt[u * 16 + arow][ak0]);
        acc[u][t] = MFMA_BF16_16x16x32(a, b, acc[u][t], 0, 0, 0);
      }
    }
    // ---- bias partials from the A fragments (kt-group-0 blocks only)
    if (db_off >= 0 && blockIdx.y == 0) {
#pragma unroll
      for (int u = 0; u < NTB; ++u) {
        const bf16x8 a =
            *reinterpret_cast<const bf16x8*>(&dZt[u * 16 + arow][ak0]);
#pragma unroll
        for (int j = 0; j < 8; ++j) db_acc[u] += (float)a[j];
      }
    }
  }

  // ---- write this wave's slab slice
  float* out = slab + (long)slice * slab_stride;
  const int col = lane & 15;  // k within tile
  const int g = lane >> 4;
#pragma unroll
  for (int u = 0; u < NTB; ++u) {
#pragma unroll
    for (int t = 0; t < KPG; ++t) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int n = u * 16 + g * 4 + r;  // D row = n (A is dZ^T)
        if (n0 + n < N_VALID)
          out[dW_off + (long)(n0 + n) * K + k0 + t * 16 + col] = acc[u][t][r];
      }
    }
    if (db_off >= 0 && blockIdx.y == 0) {
      float d = db_acc[u];
      d += __shfl_xor(d, 16);
      d += __shfl_xor(d, 32);
      if ((lane >> 4) == 0 && n0 + u * 16 + arow < N_VALID)
        out[db_off + n0 + u * 16 + arow] = d;
    }
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
                             int kpg_override, int ntb_override,
                             void* stream) {
  hipStream_t s = (hipStream_t)stream;
  int NT = (N_VALID + 15) / 16;
  // NTB n-tiles per workgroup (X-amplification = NT/NTB); KPG picked per
  // shape so the grid spans the whole 256-CU chip (swept on MI355X,
  // tools/wgrad_sweep.py): fill beats per-WG tile depth at these shapes.
  int NTB = (NT % 4 == 0) ? 4 : 1;
  int KPG = 1;
  if (ntb_override > 0) NTB = ntb_override;
  if (kpg_override > 0) {
    KPG = kpg_override;
  } else if (NT <= 1) {
    // head grads (N<=16): swept best is KPG=4 x 16 slices = 64 WGs
    // (per-WG k-depth beats chip fill at this tiny N)
    KPG = (K % 64 == 0) ? 4 : ((K % 32 == 0) ? 2 : 1);
  } else if (K % 64 == 0 && (NT / NTB) * (K / 64) >= 16) {
    // big-K layer grads: KPG=4 puts exactly 256 WGs on the 256 CUs
    // (KPG=8 halves the grid; KPG<=2 doubles it past the fill point) --
    // 27.4 -> 20.6 us for the 256x256/S=32768 shape (tools/wgrad_sweep.py)
    KPG = 4;
  } else if (K % 32 == 0) {
    KPG = 2;
  }
  int KTG = K / (16 * KPG);
  int n_slices = WG_SLICES;
  if (const char* e = getenv("STOIX_WGRAD_SLICES")) {
    int v = atoi(e);
    if (v >= 4 && v <= WG_SLICES) n_slices = v;
  }
  while (n_slices > 4 && (S % (n_slices * 32)) != 0) n_slices >>= 1;
  dim3 grid(NT / NTB, KTG, n_slices / 4), block(256);
#define WGRAD_LAUNCH(KPGV, NTBV)                                            \
  hipLaunchKernelGGL((wgrad_kernel<KPGV, NTBV>), grid, block, 0, s,         \
                     (const bf16_t*)dZ, (const bf16_t*)X, slab, dW_off,     \
                     db_off, slab_stride, S, N_STRIDE, K, N_VALID, n_slices)
  if (NTB == 4) {
    if (KPG == 8) WGRAD_LAUNCH(8, 4);
    else if (KPG == 4) WGRAD_LAUNCH(4, 4);
    else if (KPG == 2) WGRAD_LAUNCH(2, 4);
    else WGRAD_LAUNCH(1, 4);
  } else {
    if (KPG == 8) WGRAD_LAUNCH(8, 1);
    else if (KPG == 4) WGRAD_LAUNCH(4, 1);
    else if (KPG == 2) WGRAD_LAUNCH(2, 1);
    else WGRAD_LAUNCH(1, 1);
  }
#undef WGRAD_LAUNCH
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
