// Prioritised-replay sum-tree kernels (K10 of SURVEY.md §2.9; the §7
// "sum-tree concurrent update/sample on GPU" hard part).
//
// Tree layout: flat level-order array of 2*cap floats (cap = pow2 >=
// n_items); leaves at [cap, 2*cap), root at index 1. All cursors stay on
// device so the whole Rainbow/R2D2 update (sample -> loss -> priority
// writeback) is hip-graph capturable with zero host syncs — the reference
// reaches the same property through jit-compiled flashbax
// (/root/reference/stoix/systems/q_learning/ff_rainbow.py:433-444,262-266).
//
// Update: scatter new leaf values, then repair ancestors level by level
// with an IDEMPOTENT recompute (parent = left child + right child read
// fresh). Duplicate indices are safe: the scatter picks an arbitrary
// winner among equal-priority duplicates and the recompute never
// double-counts (unlike delta-propagation with atomics). Two forms:
//   * single-workgroup kernel (n <= a few K): one launch; level barriers
//     are __syncthreads() — all writes come from this workgroup, which
//     runs on ONE CU, so its own L1 is coherent for its own reads.
//   * multi-workgroup form: scatter kernel + one fix-level launch per
//     level (launch boundaries provide the global ordering). ~20 tiny
//     launches; inside a captured graph each costs ~1.5 us boundary.
//
// Sample: stratified proportional descent — thread i draws mass
// (i + u_i) * total / n and walks root->leaf in depth steps. The upper
// tree levels are a few KB and stay L2/L1-resident, so the walk is
// latency- not bandwidth-bound; one launch replaces ~depth torch gathers.
#include "common.h"

extern "C" __global__ void sumtree_update_single_kernel(
    float* __restrict__ tree, const long* __restrict__ idx,
    const float* __restrict__ prio, int n, int cap, int depth) {
  // NB: s_barrier alone does NOT drain outstanding global stores on CDNA
  // (it waits lgkmcnt, not vmcnt) — each level's writes must be fenced
  // with a workgroup-scope fence (s_waitcnt vmcnt(0)) before the barrier
  // or the next level reads stale children.
  for (int i = threadIdx.x; i < n; i += blockDim.x) {
    long leaf = (long)cap + idx[i];
    tree[leaf] = prio[i];
  }
  __threadfence_block();
  __syncthreads();
  for (int l = 1; l <= depth; ++l) {
    // two-phase per level: gather all child sums into registers FIRST,
    // then store. A fused load-add-store loop serialises on the compiler's
    // may-alias assumption (stores to `tree` vs the next iteration's
    // loads of `tree`), costing one full memory latency per item; the
    // split keeps all gathers of a level in flight together.
    float vals[8];
    long nodes[8];
    for (int base = threadIdx.x; base < n; base += blockDim.x * 8) {
      int cnt = 0;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        int i = base + j * blockDim.x;
        if (i < n) {
          long node = ((long)cap + idx[i]) >> l;
          nodes[cnt] = node;
          vals[cnt] = tree[2 * node] + tree[2 * node + 1];
          ++cnt;
        }
      }
#pragma unroll
      for (int j = 0; j < 8; ++j)
        if (j < cnt) tree[nodes[j]] = vals[j];
    }
    __threadfence_block();
    __syncthreads();
  }
}

extern "C" __global__ void sumtree_scatter_kernel(
    float* __restrict__ tree, const long* __restrict__ idx,
    const float* __restrict__ prio, int n, int cap) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) tree[(long)cap + idx[i]] = prio[i];
}

extern "C" __global__ void sumtree_fix_level_kernel(
    float* __restrict__ tree, const long* __restrict__ idx, int n, int cap,
    int level) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  long node = ((long)cap + idx[i]) >> level;
  tree[node] = tree[2 * node] + tree[2 * node + 1];
}

extern "C" __global__ void sumtree_sample_kernel(
    const float* __restrict__ tree, const float* __restrict__ u,
    long* __restrict__ out, int n, int cap, int depth, int n_items) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  float total = tree[1];
  float seg = total / (float)n;
  float mass = ((float)i + u[i]) * seg;
  long node = 1;
  for (int l = 0; l < depth; ++l) {
    long left = 2 * node;
    float ls = tree[left];
    bool right = mass >= ls;
    mass = right ? mass - ls : mass;
    node = right ? left + 1 : left;
  }
  long item = node - (long)cap;
  if (item < 0) item = 0;
  if (item > (long)(n_items - 1)) item = n_items - 1;
  out[i] = item;
}

// ----------------------------------------------------------- host launchers

extern "C" void launch_sumtree_update(float* tree, const long* idx,
                                      const float* prio, long n, int cap,
                                      int depth, void* stream) {
  hipStream_t s = (hipStream_t)stream;
  if (n <= 0) return;
  if (n <= 8192) {  // <= 8 items/thread/level: one in-flight gather chunk
    int threads = (int)((n < 1024) ? ((n + 63) / 64) * 64 : 1024);
    if (threads < 64) threads = 64;
    hipLaunchKernelGGL(sumtree_update_single_kernel, dim3(1), dim3(threads),
                       0, s, tree, idx, prio, (int)n, cap, depth);
    return;
  }
  int threads = 256;
  int blocks = (int)((n + threads - 1) / threads);
  hipLaunchKernelGGL(sumtree_scatter_kernel, dim3(blocks), dim3(threads), 0,
                     s, tree, idx, prio, (int)n, cap);
  for (int l = 1; l <= depth; ++l) {
    hipLaunchKernelGGL(sumtree_fix_level_kernel, dim3(blocks), dim3(threads),
                       0, s, tree, idx, (int)n, cap, l);
  }
}

extern "C" void launch_sumtree_sample(const float* tree, const float* u,
                                      long* out, long n, int cap, int depth,
                                      int n_items, void* stream) {
  hipStream_t s = (hipStream_t)stream;
  if (n <= 0) return;
  int threads = 256;
  int blocks = (int)((n + threads - 1) / threads);
  hipLaunchKernelGGL(sumtree_sample_kernel, dim3(blocks), dim3(threads), 0, s,
                     tree, u, out, (int)n, cap, depth, n_items);
}
