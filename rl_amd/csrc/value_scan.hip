// Fused value-estimation scans for CDNA4 (gfx950).
//
// Implements the first-order linear recurrence  y[t] = b[t] + a[t] * y[t+1]
// (reverse time) that underlies GAE, TD(lambda), discounted returns and
// V-trace (reference math: pytorch/rl torchrl/objectives/value/functional.py
// :120 generalized_advantage_estimate, :1298 vtrace_advantage_estimate —
// re-derived here as a segmented scan, not a port).
//
// Design (MI355X):
//  * one workgroup (256 threads = 4 waves) per [B] row; grid = B rows.
//    PPO-style batches have B = n_envs (4096) >> 256 CUs, so the chip fills.
//  * the (a, b) pair composition (a1,b1)∘(a2,b2) = (a1*a2, b1 + a1*b2) is
//    scanned chunk-wise: each 256-wide chunk does a Hillis-Steele suffix
//    scan in LDS (8 rounds), then the running carry from the later chunk
//    is applied.  Work is O(T log 256) per row, all resident in LDS.
//  * inputs are read once, delta/g are computed in-register (no
//    intermediate HBM tensors — the torch path materializes 5).
//  * fp32 and bf16 I/O; accumulation always fp32.
//
// Validated bit-for-bit (fp32) against functional.py oracles in
// tests/test_ops.py.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WG 256

namespace {

template <typename T>
__device__ __forceinline__ float ld(const T* p, long i);

template <>
__device__ __forceinline__ float ld<float>(const float* p, long i) {
  return p[i];
}
template <>
__device__ __forceinline__ float ld<__hip_bfloat16>(const __hip_bfloat16* p, long i) {
  return __bfloat162float(p[i]);
}

template <typename T>
__device__ __forceinline__ void st(T* p, long i, float v);

template <>
__device__ __forceinline__ void st<float>(float* p, long i, float v) {
  p[i] = v;
}
template <>
__device__ __forceinline__ void st<__hip_bfloat16>(__hip_bfloat16* p, long i, float v) {
  p[i] = __float2bfloat16(v);
}

// Suffix-scan a chunk of (a,b) pairs held in LDS, in place.
// After the scan, (a_s[i], b_s[i]) composes elements i..len-1 of the chunk.
__device__ __forceinline__ void chunk_suffix_scan(float* a_s, float* b_s, int len) {
  const int tid = threadIdx.x;
  for (int off = 1; off < len; off <<= 1) {
    float na = 1.f, nb = 0.f;
    bool active = (tid + off) < len && tid < len;
    if (active) {
      na = a_s[tid + off];
      nb = b_s[tid + off];
    }
    __syncthreads();
    if (active) {
      // (a_t, b_t) ∘ (a_next, b_next)
      b_s[tid] = b_s[tid] + a_s[tid] * nb;
      a_s[tid] = a_s[tid] * na;
    }
    __syncthreads();
  }
}

// ---------------------------------------------------------------------------
// GAE: delta = r + gamma*nt*V' - V ;  adv[t] = delta[t] + gamma*lmbda*nd[t]*adv[t+1]
//      vtarget = adv + V
// ---------------------------------------------------------------------------
template <typename T>
__global__ void gae_kernel(
    const T* __restrict__ reward,
    const T* __restrict__ value,
    const T* __restrict__ next_value,
    const bool* __restrict__ done,
    const bool* __restrict__ terminated,
    T* __restrict__ adv,
    T* __restrict__ vtarget,
    const long T_len,
    const float gamma,
    const float lmbda) {
  __shared__ float a_s[WG];
  __shared__ float b_s[WG];
  const long row = blockIdx.x;
  const long base = row * T_len;
  const int tid = threadIdx.x;

  float carry = 0.f;  // adv at chunk_start of the LATER chunk
  // walk chunks from the tail
  const long n_chunks = (T_len + WG - 1) / WG;
  for (long c = n_chunks - 1; c >= 0; --c) {
    const long start = c * WG;
    const int len = (int)min((long)WG, T_len - start);
    const long t = start + tid;
    float a = 1.f, b = 0.f;
    if (tid < len) {
      const float r = ld(reward, base + t);
      const float v = ld(value, base + t);
      const float nv = ld(next_value, base + t);
      const float nt = terminated[base + t] ? 0.f : 1.f;
      const float nd = done[base + t] ? 0.f : 1.f;
      b = r + gamma * nt * nv - v;        // delta
      a = gamma * lmbda * nd;
    }
    a_s[tid] = a;
    b_s[tid] = b;
    __syncthreads();
    chunk_suffix_scan(a_s, b_s, len);
    if (tid < len) {
      const float y = b_s[tid] + a_s[tid] * carry;
      const float v = ld(value, base + t);
      st(adv, base + t, y);
      st(vtarget, base + t, y + v);
    }
    __syncthreads();
    if (tid == 0) {
      // adv at this chunk's first element becomes the next carry
      carry = b_s[0] + a_s[0] * carry;
      a_s[0] = carry;
    }
    __syncthreads();
    carry = a_s[0];
    __syncthreads();
  }
}

// ---------------------------------------------------------------------------
// Generic reverse scan: y[t] = b[t] + a[t]*y[t+1]  (used by td-lambda,
// reward2go and any custom recurrence; a and b precomputed)
// ---------------------------------------------------------------------------
template <typename T>
__global__ void revscan_kernel(
    const T* __restrict__ a_in,
    const T* __restrict__ b_in,
    T* __restrict__ out,
    const long T_len) {
  __shared__ float a_s[WG];
  __shared__ float b_s[WG];
  const long row = blockIdx.x;
  const long base = row * T_len;
  const int tid = threadIdx.x;

  float carry = 0.f;
  const long n_chunks = (T_len + WG - 1) / WG;
  for (long c = n_chunks - 1; c >= 0; --c) {
    const long start = c * WG;
    const int len = (int)min((long)WG, T_len - start);
    const long t = start + tid;
    float a = 1.f, b = 0.f;
    if (tid < len) {
      a = ld(a_in, base + t);
      b = ld(b_in, base + t);
    }
    a_s[tid] = a;
    b_s[tid] = b;
    __syncthreads();
    chunk_suffix_scan(a_s, b_s, len);
    if (tid < len) {
      st(out, base + t, b_s[tid] + a_s[tid] * carry);
    }
    __syncthreads();
    if (tid == 0) {
      carry = b_s[0] + a_s[0] * carry;
      a_s[0] = carry;
    }
    __syncthreads();
    carry = a_s[0];
    __syncthreads();
  }
}

// ---------------------------------------------------------------------------
// V-trace: fused ratio clamp + delta + scan + advantage epilogue.
//   rho  = min(exp(log_pi - log_mu), rho_thresh)
//   c    = min(exp(log_pi - log_mu), c_thresh)
//   delta= rho * (r + gamma*nt*V' - V)
//   vmv[t] = delta[t] + gamma*nd[t]*c[t]*vmv[t+1]      (scan)
//   vs   = vmv + V
//   vs_next[t] = nd[t] ? vs[t+1] (or V'[T-1]) : V'[t]
//   adv  = rho * (r + gamma*nt*vs_next - V)
// ---------------------------------------------------------------------------
template <typename T>
__global__ void vtrace_kernel(
    const T* __restrict__ log_pi,
    const T* __restrict__ log_mu,
    const T* __restrict__ reward,
    const T* __restrict__ value,
    const T* __restrict__ next_value,
    const bool* __restrict__ done,
    const bool* __restrict__ terminated,
    T* __restrict__ adv,
    T* __restrict__ vs_out,
    const long T_len,
    const float gamma,
    const float rho_thresh,
    const float c_thresh) {
  __shared__ float a_s[WG];
  __shared__ float b_s[WG];
  __shared__ float vs_next_first;  // vs at chunk_start of the LATER chunk
  const long row = blockIdx.x;
  const long base = row * T_len;
  const int tid = threadIdx.x;

  float carry = 0.f;
  const long n_chunks = (T_len + WG - 1) / WG;
  for (long c = n_chunks - 1; c >= 0; --c) {
    const long start = c * WG;
    const int len = (int)min((long)WG, T_len - start);
    const long t = start + tid;
    float a = 1.f, b = 0.f;
    float rho = 0.f;
    if (tid < len) {
      const float ratio = __expf(ld(log_pi, base + t) - ld(log_mu, base + t));
      rho = fminf(ratio, rho_thresh);
      const float cc = fminf(ratio, c_thresh);
      const float r = ld(reward, base + t);
      const float v = ld(value, base + t);
      const float nv = ld(next_value, base + t);
      const float nt = terminated[base + t] ? 0.f : 1.f;
      const float nd = done[base + t] ? 0.f : 1.f;
      b = rho * (r + gamma * nt * nv - v);
      a = gamma * nd * cc;
    }
    a_s[tid] = a;
    b_s[tid] = b;
    __syncthreads();
    chunk_suffix_scan(a_s, b_s, len);
    float vs = 0.f;
    if (tid < len) {
      const float v = ld(value, base + t);
      vs = b_s[tid] + a_s[tid] * carry + v;
      st(vs_out, base + t, vs);
    }
    __syncthreads();
    if (tid == 0) {
      carry = b_s[0] + a_s[0] * carry;
      a_s[0] = carry;
    }
    __syncthreads();
    carry = a_s[0];
    __syncthreads();
  }
  // second pass: advantage epilogue (vs now resident in L2)
  for (long t0 = 0; t0 < T_len; t0 += WG) {
    const long t = t0 + tid;
    if (t < T_len) {
      const float ratio = __expf(ld(log_pi, base + t) - ld(log_mu, base + t));
      const float rho = fminf(ratio, rho_thresh);
      const float r = ld(reward, base + t);
      const float v = ld(value, base + t);
      const float nv = ld(next_value, base + t);
      const float nt = terminated[base + t] ? 0.f : 1.f;
      const float nd = done[base + t] ? 0.f : 1.f;
      float vsn;
      if (done[base + t] || t == T_len - 1) {
        vsn = nv;
      } else {
        vsn = ld(vs_out, base + t + 1);
      }
      // at done boundaries bootstrap from the true next value
      vsn = nd * vsn + (1.f - nd) * nv;
      if (t == T_len - 1) vsn = nv;
      st(adv, base + t, rho * (r + gamma * nt * vsn - v));
    }
  }
}

}  // namespace

// ---------------------------------------------------------------------------
// C-visible launchers (stream-ordered; no syncs here)
// ---------------------------------------------------------------------------
extern "C" {

void launch_gae_f32(const float* r, const float* v, const float* nv,
                    const bool* d, const bool* tm, float* adv, float* vt,
                    long B, long T, float gamma, float lmbda, void* stream) {
  hipLaunchKernelGGL(gae_kernel<float>, dim3(B), dim3(WG), 0,
                     (hipStream_t)stream, r, v, nv, d, tm, adv, vt, T, gamma,
                     lmbda);
}

void launch_gae_bf16(const void* r, const void* v, const void* nv,
                     const bool* d, const bool* tm, void* adv, void* vt,
                     long B, long T, float gamma, float lmbda, void* stream) {
  hipLaunchKernelGGL(gae_kernel<__hip_bfloat16>, dim3(B), dim3(WG), 0,
                     (hipStream_t)stream, (const __hip_bfloat16*)r,
                     (const __hip_bfloat16*)v, (const __hip_bfloat16*)nv, d, tm,
                     (__hip_bfloat16*)adv, (__hip_bfloat16*)vt, T, gamma, lmbda);
}

void launch_revscan_f32(const float* a, const float* b, float* y, long B,
                        long T, void* stream) {
  hipLaunchKernelGGL(revscan_kernel<float>, dim3(B), dim3(WG), 0,
                     (hipStream_t)stream, a, b, y, T);
}

void launch_revscan_bf16(const void* a, const void* b, void* y, long B, long T,
                         void* stream) {
  hipLaunchKernelGGL(revscan_kernel<__hip_bfloat16>, dim3(B), dim3(WG), 0,
                     (hipStream_t)stream, (const __hip_bfloat16*)a,
                     (const __hip_bfloat16*)b, (__hip_bfloat16*)y, T);
}

void launch_vtrace_f32(const float* lp, const float* lm, const float* r,
                       const float* v, const float* nv, const bool* d,
                       const bool* tm, float* adv, float* vs, long B, long T,
                       float gamma, float rho_th, float c_th, void* stream) {
  hipLaunchKernelGGL(vtrace_kernel<float>, dim3(B), dim3(WG), 0,
                     (hipStream_t)stream, lp, lm, r, v, nv, d, tm, adv, vs, T,
                     gamma, rho_th, c_th);
}

}  // extern "C"
