// Fused optimiser kernels (K8/K9 of SURVEY.md §2.9).
//
// The reference's optax chain(clip_by_global_norm, adam) + apply_updates +
// incremental_update (ff_ppo.py:456-463, ff_dqn.py:207-209) become three
// kernels over ONE flat parameter buffer: a grid-stride squared-norm
// reduction (one atomicAdd per block), a fused clip+Adam update reading the
// norm from device memory (no host sync inside the graph), and a polyak
// target update. All are memory-bound: loads are float4-vectorised.
#include "common.h"

// partial squared-norm: block-level shuffle reduction + one atomic per block
extern "C" __global__ void grad_sqnorm_kernel(
    const float* __restrict__ grad, float* __restrict__ out, long n) {
  long tid = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  float acc = 0.0f;
  long nvec = n / 4;  // flat buffers are 16B-aligned (torch allocator)
  const float4* g4 = reinterpret_cast<const float4*>(grad);
  for (long v = tid; v < nvec; v += stride) {
    float4 g = g4[v];
    acc += g.x * g.x + g.y * g.y + g.z * g.z + g.w * g.w;
  }
  for (long j = nvec * 4 + tid; j < n; j += stride) acc += grad[j] * grad[j];
  // wave reduce (wave64)
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) acc += __shfl_down(acc, off);
  __shared__ float warp_sums[16];
  int lane = threadIdx.x & 63;
  int wid = threadIdx.x >> 6;
  if (lane == 0) warp_sums[wid] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    float s = 0.0f;
    int nw = (blockDim.x + 63) / 64;
    for (int w = 0; w < nw; ++w) s += warp_sums[w];
    atomicAdd(out, s);
  }
}

// fused global-norm clip + Adam. sqnorm is a 1-element device buffer written
// by grad_sqnorm_kernel; step_t a device int64 incremented by the caller
// kernel-side (adam_inc) so the whole sequence is graph-replayable.
extern "C" __global__ void adam_update_kernel(
    float* __restrict__ param,
    const float* __restrict__ grad,
    float* __restrict__ exp_avg,
    float* __restrict__ exp_avg_sq,
    const float* __restrict__ sqnorm,   // [1]
    const long* __restrict__ step_t,    // [1] (already incremented)
    long n, float lr, float beta1, float beta2, float eps, float max_norm) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  float clip = 1.0f;
  if (max_norm > 0.0f) {
    float norm = sqrtf(*sqnorm);
    clip = (norm > max_norm) ? (max_norm / (norm + 1e-6f)) : 1.0f;
  }
  long t = *step_t;
  float bc1 = 1.0f - powf(beta1, (float)t);
  float bc2 = 1.0f - powf(beta2, (float)t);
  for (; i < n; i += stride) {
    float g = grad[i] * clip;
    float m = exp_avg[i] = beta1 * exp_avg[i] + (1.0f - beta1) * g;
    float v = exp_avg_sq[i] = beta2 * exp_avg_sq[i] + (1.0f - beta2) * g * g;
    float mhat = m / bc1;
    float vhat = v / bc2;
    param[i] -= lr * mhat / (sqrtf(vhat) + eps);
  }
}

extern "C" __global__ void adam_prologue_kernel(float* __restrict__ sqnorm,
                                                long* __restrict__ step_t) {
  if (blockIdx.x == 0 && threadIdx.x == 0) {
    *sqnorm = 0.0f;
    *step_t += 1;
  }
}

// bf16-gradient variants for the fused update path (stoix_amd/ops/csrc/
// mlp.hip): the minibatch backward produces bf16 grads written straight
// into one flat buffer by the wgrad GEMMs; the RCCL all-reduce runs on the
// bf16 buffer; Adam reads bf16 grads, keeps the fp32 master param, and
// mirrors the updated param to bf16 (the GEMM/rollout-kernel weights) in
// the same pass — no separate cast kernels anywhere in the loop.
#include <hip/hip_bf16.h>
typedef __bf16 opt_bf16;
typedef __bf16 opt_bf16x8 __attribute__((ext_vector_type(8)));

extern "C" __global__ void grad_sqnorm_bf16_kernel(
    const opt_bf16* __restrict__ grad, float* __restrict__ out, long n) {
  long tid = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  float acc = 0.0f;
  long nvec = n / 8;
  const opt_bf16x8* g8 = reinterpret_cast<const opt_bf16x8*>(grad);
  for (long v = tid; v < nvec; v += stride) {
    opt_bf16x8 g = g8[v];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float x = (float)g[j];
      acc += x * x;
    }
  }
  for (long j = nvec * 8 + tid; j < n; j += stride) {
    float x = (float)grad[j];
    acc += x * x;
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) acc += __shfl_down(acc, off);
  __shared__ float warp_sums[16];
  int lane = threadIdx.x & 63;
  int wid = threadIdx.x >> 6;
  if (lane == 0) warp_sums[wid] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    float s = 0.0f;
    int nw = (blockDim.x + 63) / 64;
    for (int w = 0; w < nw; ++w) s += warp_sums[w];
    atomicAdd(out, s);
  }
}

extern "C" __global__ void adam_update_bf16_kernel(
    float* __restrict__ param, const opt_bf16* __restrict__ grad,
    float* __restrict__ exp_avg, float* __restrict__ exp_avg_sq,
    const float* __restrict__ sqnorm, const long* __restrict__ step_t,
    opt_bf16* __restrict__ param_bf16, long n, float lr, float beta1,
    float beta2, float eps, float max_norm, float grad_scale) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  float clip = grad_scale;
  if (max_norm > 0.0f) {
    float norm = sqrtf(*sqnorm) * grad_scale;
    if (norm > max_norm) clip = grad_scale * max_norm / (norm + 1e-6f);
  }
  long t = *step_t;
  float bc1 = 1.0f - powf(beta1, (float)t);
  float bc2 = 1.0f - powf(beta2, (float)t);
  for (; i < n; i += stride) {
    float g = (float)grad[i] * clip;
    float m = exp_avg[i] = beta1 * exp_avg[i] + (1.0f - beta1) * g;
    float v = exp_avg_sq[i] = beta2 * exp_avg_sq[i] + (1.0f - beta2) * g * g;
    float p = param[i] - lr * (m / bc1) / (sqrtf(v / bc2) + eps);
    param[i] = p;
    if (param_bf16) param_bf16[i] = (opt_bf16)p;
  }
}

extern "C" void launch_fused_adam_bf16(float* param, const void* grad,
                                       float* exp_avg, float* exp_avg_sq,
                                       float* sqnorm, long* step_t,
                                       void* param_bf16, long n, float lr,
                                       float beta1, float beta2, float eps,
                                       float max_norm, float grad_scale,
                                       int do_prologue, void* stream) {
  hipStream_t s = (hipStream_t)stream;
  // the prologue (zero sqnorm, step+1) can ride along in slab_reduce
  // (wgrad.hip) when that kernel precedes this one in the same stream
  if (do_prologue)
    hipLaunchKernelGGL(adam_prologue_kernel, dim3(1), dim3(1), 0, s, sqnorm,
                       step_t);
  int threads = 256;
  long want = (n / 8 + threads - 1) / threads;
  int blocks = (int)(want < 2048 ? (want > 0 ? want : 1) : 2048);
  if (max_norm > 0.0f) {
    hipLaunchKernelGGL(grad_sqnorm_bf16_kernel, dim3(blocks), dim3(threads),
                       0, s, (const opt_bf16*)grad, sqnorm, n);
  }
  long want2 = (n + threads - 1) / threads;
  int blocks2 = (int)(want2 < 2048 ? (want2 > 0 ? want2 : 1) : 2048);
  hipLaunchKernelGGL(adam_update_bf16_kernel, dim3(blocks2), dim3(threads), 0,
                     s, param, (const opt_bf16*)grad, exp_avg, exp_avg_sq,
                     sqnorm, step_t, (opt_bf16*)param_bf16, n, lr, beta1,
                     beta2, eps, max_norm, grad_scale);
}

// polyak: target <- tau * online + (1 - tau) * target (ff_dqn.py:207-209)
extern "C" __global__ void polyak_kernel(const float* __restrict__ online,
                                         float* __restrict__ target,
                                         long n, float tau) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (; i < n; i += stride)
    target[i] = tau * online[i] + (1.0f - tau) * target[i];
}

// ----------------------------------------------------------- host launchers
extern "C" void launch_fused_adam(float* param, const float* grad,
                                  float* exp_avg, float* exp_avg_sq,
                                  float* sqnorm, long* step_t, long n,
                                  float lr, float beta1, float beta2,
                                  float eps, float max_norm, void* stream) {
  hipStream_t s = (hipStream_t)stream;
  hipLaunchKernelGGL(adam_prologue_kernel, dim3(1), dim3(1), 0, s, sqnorm, step_t);
  int threads = 256;
  long want = (n / 4 + threads - 1) / threads;
  int blocks = (int)(want < 2048 ? (want > 0 ? want : 1) : 2048);
  if (max_norm > 0.0f) {
    hipLaunchKernelGGL(grad_sqnorm_kernel, dim3(blocks), dim3(threads), 0, s,
                       grad, sqnorm, n);
  }
  long want2 = (n + threads - 1) / threads;
  int blocks2 = (int)(want2 < 2048 ? (want2 > 0 ? want2 : 1) : 2048);
  hipLaunchKernelGGL(adam_update_kernel, dim3(blocks2), dim3(threads), 0, s,
                     param, grad, exp_avg, exp_avg_sq, sqnorm, step_t, n, lr,
                     beta1, beta2, eps, max_norm);
}

extern "C" void launch_polyak(const float* online, float* target, long n,
                              float tau, void* stream) {
  int threads = 256;
  long want = (n + threads - 1) / threads;
  int blocks = (int)(want < 2048 ? (want > 0 ? want : 1) : 2048);
  hipLaunchKernelGGL(polyak_kernel, dim3(blocks), dim3(threads), 0,
                     (hipStream_t)stream, online, target, n, tau);
}
