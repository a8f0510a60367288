// Device segment-tree kernels for prioritized replay on MI355X.
//
// Capability of the reference's CUDA trees
// (pytorch/rl torchrl/csrc/cuda_segment_tree.cu: SetLeavesKernel:27,
// RecomputeLevelKernel:41, QueryKernel:52, ScanLowerBoundKernel:76)
// re-designed for CDNA4 — fresh implementation, not a port:
//
//  * scan_lower_bound: one thread per sample walks root→leaf; the top
//    LDS_TOP nodes (shared by every descent path) are staged in LDS per
//    workgroup, so the hottest levels never leave the CU.
//  * update: host pre-dedupes duplicate indices (last-writer-wins — the
//    ordering semantics the reference gets by serializing leaf writes in
//    one thread), then TWO kernels run the parallel path recompute:
//      pass 1 marks every ancestor of every updated leaf with an atomic
//      counter; pass 2 ascends again — at each node the LAST arriving
//      path (atomicSub reaches 0) recomputes the node from its now-final
//      children and continues, earlier arrivals retire.  Each node is
//      recomputed exactly once, after all contributing subtrees are final.
//    All counters return to zero after pass 2, so the workspace needs no
//    re-zeroing between calls.
//
// Correctness under XCD non-coherence (guide §6 G16): all cross-workgroup
// traffic is device-scope atomics + __threadfence().

#include <hip/hip_runtime.h>

#define WG 256
#define LDS_TOP 1024

namespace {

__global__ void tree_scan_kernel(const double* __restrict__ tree,
                                 const double* __restrict__ mass,
                                 long* __restrict__ out, const long n,
                                 const long size, const long capacity) {
  __shared__ double top[LDS_TOP];
  for (int i = threadIdx.x; i < LDS_TOP; i += WG) {
    top[i] = (i >= 1 && i < 2 * size) ? tree[i] : 0.0;
  }
  __syncthreads();
  const long g = blockIdx.x * (long)WG + threadIdx.x;
  if (g >= n) return;
  double rem = mass[g];
  long node = 1;
  while (node < size) {
    const long left = 2 * node;
    const double lv = (left < LDS_TOP) ? top[left] : tree[left];
    if (rem >= lv) {
      rem -= lv;
      node = left + 1;
    } else {
      node = left;
    }
  }
  long leaf = node - size;
  out[g] = leaf < capacity ? leaf : capacity - 1;
}

// pass 0: scatter leaves (indices pre-deduped on the host side)
__global__ void tree_set_leaves_kernel(double* __restrict__ sum_tree,
                                       double* __restrict__ min_tree,
                                       const long* __restrict__ index,
                                       const double* __restrict__ value,
                                       const long n, const long size,
                                       const int with_min) {
  const long g = blockIdx.x * (long)WG + threadIdx.x;
  if (g >= n) return;
  const long node = size + index[g];
  sum_tree[node] = value[g];
  if (with_min) min_tree[node] = value[g];
}

// pass 1: mark arrival counts.  FIRST arrival at a node continues upward,
// later arrivals stop — so cnt[P] == number of distinct child paths that
// will arrive at P during resolve (1 or 2), and every node on any updated
// path is marked exactly once per arriving path.
__global__ void tree_mark_kernel(int* __restrict__ cnt,
                                 const long* __restrict__ index, const long n,
                                 const long size) {
  const long g = blockIdx.x * (long)WG + threadIdx.x;
  if (g >= n) return;
  long node = (size + index[g]) >> 1;
  while (node >= 1) {
    const int prev = atomicAdd(&cnt[node], 1);
    if (prev > 0) return;  // someone already marked this node and above
    node >>= 1;
  }
}

// pass 2: LAST arrival recomputes and continues; earlier arrivals retire.
// Mirrors the mark gating, so exactly cnt[P] threads decrement each node
// and every counter returns to zero.
__global__ void tree_resolve_kernel(double* __restrict__ sum_tree,
                                    double* __restrict__ min_tree,
                                    int* __restrict__ cnt,
                                    const long* __restrict__ index,
                                    const long n, const long size,
                                    const int with_min) {
  const long g = blockIdx.x * (long)WG + threadIdx.x;
  if (g >= n) return;
  long node = (size + index[g]) >> 1;
  while (node >= 1) {
    const int prev = atomicSub(&cnt[node], 1);
    if (prev > 1) return;  // the other child path finishes this node
    __threadfence();  // acquire: sibling subtree writes are now visible
    const double l = sum_tree[2 * node];
    const double r = sum_tree[2 * node + 1];
    sum_tree[node] = l + r;
    if (with_min) {
      const double lm = min_tree[2 * node];
      const double rm = min_tree[2 * node + 1];
      min_tree[node] = lm < rm ? lm : rm;
    }
    __threadfence();  // release: make this node visible before parent's atomic
    node >>= 1;
  }
}

}  // namespace

extern "C" {

void launch_tree_scan_f64(const double* tree, const double* mass, long* out,
                          long n, long size, long capacity, void* stream) {
  const long blocks = (n + WG - 1) / WG;
  hipLaunchKernelGGL(tree_scan_kernel, dim3(blocks), dim3(WG), 0,
                     (hipStream_t)stream, tree, mass, out, n, size, capacity);
}

void launch_tree_update_f64(double* sum_tree, double* min_tree, int* cnt,
                            const long* index, const double* value, long n,
                            long size, int with_min, void* stream) {
  const long blocks = (n + WG - 1) / WG;
  hipLaunchKernelGGL(tree_set_leaves_kernel, dim3(blocks), dim3(WG), 0,
                     (hipStream_t)stream, sum_tree, min_tree, index, value, n,
                     size, with_min);
  hipLaunchKernelGGL(tree_mark_kernel, dim3(blocks), dim3(WG), 0,
                     (hipStream_t)stream, cnt, index, n, size);
  hipLaunchKernelGGL(tree_resolve_kernel, dim3(blocks), dim3(WG), 0,
                     (hipStream_t)stream, sum_tree, min_tree, cnt, index, n,
                     size, with_min);
}

}  // extern "C"
