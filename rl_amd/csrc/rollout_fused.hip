// Whole-rollout mega-kernel (CDNA4, gfx950): the ENTIRE T-step
// PPO rollout in ONE launch.
//
// Insight: env rows are independent — no cross-row (and hence no
// cross-workgroup) dependency exists anywhere in the rollout.  Each
// workgroup owns R_TILE rows and loops all T steps locally:
//
//   per step t:
//     store obs[r, t]                 (pre-step observation)
//     actor: MLP(tanh)x2 + heads + TanhNormal sample + log-prob
//            (weights staged once in LDS, eps pre-generated [T, B, A])
//     env:   s' = tanh(s @ A + a @ Bm); reward = s'[0] - 0.1|a|^2;
//            done = (++step >= max_steps); carried state auto-resets
//            from noise[t] (terminal obs still goes to the store)
//     store action/log-prob/next-obs/reward/done at [r, t]
//
// State lives in LDS across the whole loop; the persistent [B, S]
// state and [B] step counters are read once and written once.  This
// removes the per-step launch/latency floor of the 2-kernels-per-step
// rollout (64 dependent launches at T=64) — the rollout becomes one
// kernel + two randn calls per iteration.
//
// Numerics match the fused_actor + synthetic_env_step pair bit-for-bit
// (same math, same order); validated in tests/test_ops.py.

#include <hip/hip_bf16.h>
#include <hip/hip_runtime.h>

#define RO_THREADS 256
#define RO_ROWS 4

namespace {

__device__ __forceinline__ float softplusf_(float x) {
  if (x > 20.f) return x;
  return log1pf(__expf(x));
}

__global__ void __launch_bounds__(RO_THREADS) fused_rollout_kernel(
    float* __restrict__ state,       // [B, S] in/out (carried)
    float* __restrict__ step_ct,     // [B] in/out (env step counter)
    const float* __restrict__ w1,    // [H1, O]
    const float* __restrict__ b1,    // [H1]
    const float* __restrict__ w2,    // [H2, H1]
    const float* __restrict__ b2,    // [H2]
    const float* __restrict__ w3,    // [2A, H2]
    const float* __restrict__ b3,    // [2A]
    const float* __restrict__ Amat,  // [S, S]
    const float* __restrict__ Bmat,  // [Aact, S]
    const float* __restrict__ eps,   // [T, B, Aact]
    const float* __restrict__ noise, // [T, B, S] (reset states)
    float* __restrict__ st_obs,      // [B, T, S]
    float* __restrict__ st_act,      // [B, T, Aact]
    float* __restrict__ st_logp,     // [B, T]
    float* __restrict__ st_nobs,     // [B, T, S]
    float* __restrict__ st_rew,      // [B, T, 1]
    bool* __restrict__ st_done,      // [B, T, 1]
    const int B, const int S, const int H1, const int H2, const int Aact,
    const int T, const float max_steps, const float inv_softplus_bias,
    const float scale_lb) {
  extern __shared__ float smem[];
  const int w1s = S | 1, w2s = H1 | 1, w3s = H2 | 1, as = S | 1;
  float* s_w1 = smem;
  float* s_b1 = s_w1 + H1 * w1s;
  float* s_w2 = s_b1 + H1;
  float* s_b2 = s_w2 + H2 * w2s;
  float* s_w3 = s_b2 + H2;
  float* s_b3 = s_w3 + 2 * Aact * w3s;
  float* s_A = s_b3 + 2 * Aact;            // [S, as]
  float* s_B = s_A + S * as;               // [Aact, as]
  const int bufw = max(max(S, H1), max(H2, 2 * Aact));
  float* s_state = s_B + Aact * as;        // [RO_ROWS, as]
  float* s_act = s_state + RO_ROWS * as;   // [RO_ROWS, Aact]
  float* buf_a = s_act + RO_ROWS * Aact;   // [RO_ROWS, bufw]
  float* buf_b = buf_a + RO_ROWS * bufw;   // [RO_ROWS, bufw]
  float* s_ct = buf_b + RO_ROWS * bufw;    // [RO_ROWS]

  const int tid = threadIdx.x;
  const int row0 = blockIdx.x * RO_ROWS;
  const int rows = min(RO_ROWS, B - row0);

  // one-time staging
  for (int i = tid; i < H1 * S; i += RO_THREADS) s_w1[(i / S) * w1s + i % S] = w1[i];
  for (int i = tid; i < H1; i += RO_THREADS) s_b1[i] = b1[i];
  for (int i = tid; i < H2 * H1; i += RO_THREADS) s_w2[(i / H1) * w2s + i % H1] = w2[i];
  for (int i = tid; i < H2; i += RO_THREADS) s_b2[i] = b2[i];
  for (int i = tid; i < 2 * Aact * H2; i += RO_THREADS) s_w3[(i / H2) * w3s + i % H2] = w3[i];
  for (int i = tid; i < 2 * Aact; i += RO_THREADS) s_b3[i] = b3[i];
  for (int i = tid; i < S * S; i += RO_THREADS) s_A[(i / S) * as + i % S] = Amat[i];
  for (int i = tid; i < Aact * S; i += RO_THREADS) s_B[(i / S) * as + i % S] = Bmat[i];
  for (int i = tid; i < rows * S; i += RO_THREADS)
    s_state[(i / S) * as + i % S] = state[(size_t)(row0 + i / S) * S + i % S];
  for (int r = tid; r < rows; r += RO_THREADS) s_ct[r] = step_ct[row0 + r];
  __syncthreads();

  const float LOG_SQRT_2PI = 0.9189385332046727f;
  const float LOG2 = 0.6931471805599453f;
  const float lim = 1.0f - 1.1920929e-7f;

  for (int t = 0; t < T; ++t) {
    // store pre-step obs
    for (int i = tid; i < rows * S; i += RO_THREADS) {
      const int r = i / S, j = i % S;
      st_obs[((size_t)(row0 + r) * T + t) * S + j] = s_state[r * as + j];
    }
    // actor layer 1
    for (int i = tid; i < rows * H1; i += RO_THREADS) {
      const int r = i / H1, j = i % H1;
      float acc = s_b1[j];
      const float* in = &s_state[r * as];
      const float* wr = &s_w1[j * w1s];
#pragma unroll 4
      for (int k = 0; k < S; ++k) acc += in[k] * wr[k];
      buf_a[r * bufw + j] = tanhf(acc);
    }
    __syncthreads();
    // actor layer 2
    for (int i = tid; i < rows * H2; i += RO_THREADS) {
      const int r = i / H2, j = i % H2;
      float acc = s_b2[j];
      const float* in = &buf_a[r * bufw];
      const float* wr = &s_w2[j * w2s];
#pragma unroll 8
      for (int k = 0; k < H1; ++k) acc += in[k] * wr[k];
      buf_b[r * bufw + j] = tanhf(acc);
    }
    __syncthreads();
    // heads
    for (int i = tid; i < rows * 2 * Aact; i += RO_THREADS) {
      const int r = i / (2 * Aact), j = i % (2 * Aact);
      float acc = s_b3[j];
      const float* in = &buf_b[r * bufw];
      const float* wr = &s_w3[j * w3s];
#pragma unroll 8
      for (int k = 0; k < H2; ++k) acc += in[k] * wr[k];
      buf_a[r * bufw + j] = acc;
    }
    __syncthreads();
    // sample + per-element log-prob
    for (int i = tid; i < rows * Aact; i += RO_THREADS) {
      const int r = i / Aact, a = i % Aact;
      const float loc = buf_a[r * bufw + a];
      float scale = softplusf_(buf_a[r * bufw + Aact + a] + inv_softplus_bias);
      scale = fmaxf(scale, scale_lb);
      const float e = eps[((size_t)t * B + row0 + r) * Aact + a];
      const float u = loc + scale * e;
      float act = tanhf(u);
      act = fminf(fmaxf(act, -lim), lim);
      // env consumes clamp(-1,1) of the action — same value here
      s_act[r * Aact + a] = act;
      st_act[((size_t)(row0 + r) * T + t) * Aact + a] = act;
      buf_b[r * bufw + a] =
          -0.5f * e * e - __logf(scale) - LOG_SQRT_2PI
          - 2.0f * (LOG2 - u - softplusf_(-2.0f * u));
    }
    __syncthreads();
    // per-row log-prob sum
    for (int r = tid; r < rows; r += RO_THREADS) {
      float sum = 0.f;
      for (int a = 0; a < Aact; ++a) sum += buf_b[r * bufw + a];
      st_logp[(size_t)(row0 + r) * T + t] = sum;
    }
    // env transition into buf_a (next state, pre-reset)
    for (int i = tid; i < rows * S; i += RO_THREADS) {
      const int r = i / S, j = i % S;
      float acc = 0.f;
      const float* sr = &s_state[r * as];
#pragma unroll 4
      for (int k = 0; k < S; ++k) acc += sr[k] * s_A[k * as + j];
      const float* ar = &s_act[r * Aact];
#pragma unroll
      for (int k = 0; k < Aact; ++k) acc += ar[k] * s_B[k * as + j];
      buf_a[r * bufw + j] = tanhf(acc);
    }
    __syncthreads();
    // bookkeeping + store + reset-carry
    for (int i = tid; i < rows * S; i += RO_THREADS) {
      const int r = i / S, j = i % S;
      const bool trunc = (s_ct[r] + 1.f) >= max_steps;
      const float ns = buf_a[r * bufw + j];
      st_nobs[((size_t)(row0 + r) * T + t) * S + j] = ns;
      const float carry =
          trunc ? noise[((size_t)t * B + row0 + r) * S + j] : ns;
      buf_b[r * bufw + j] = carry;  // staged; committed after sync
      if (j == 0) {
        float ctrl = 0.f;
#pragma unroll
        for (int k = 0; k < Aact; ++k) ctrl += s_act[r * Aact + k] * s_act[r * Aact + k];
        st_rew[(size_t)(row0 + r) * T + t] = ns - 0.1f * ctrl;
        st_done[(size_t)(row0 + r) * T + t] = trunc;
      }
    }
    __syncthreads();
    for (int i = tid; i < rows * S; i += RO_THREADS) {
      const int r = i / S, j = i % S;
      s_state[r * as + j] = buf_b[r * bufw + j];
    }
    for (int r = tid; r < rows; r += RO_THREADS) {
      const bool trunc = (s_ct[r] + 1.f) >= max_steps;
      s_ct[r] = trunc ? 0.f : s_ct[r] + 1.f;
    }
    __syncthreads();
  }

  // commit carried state + counters
  for (int i = tid; i < rows * S; i += RO_THREADS)
    state[(size_t)(row0 + i / S) * S + i % S] = s_state[(i / S) * as + i % S];
  for (int r = tid; r < rows; r += RO_THREADS) step_ct[row0 + r] = s_ct[r];
}

}  // namespace

extern "C" int fused_rollout_lds_bytes(int S, int H1, int H2, int Aact) {
  const int bufw = max(max(S, H1), max(H2, 2 * Aact));
  return (int)sizeof(float) *
         (H1 * (S | 1) + H1 + H2 * (H1 | 1) + H2 + 2 * Aact * (H2 | 1) +
          2 * Aact + S * (S | 1) + Aact * (S | 1) + RO_ROWS * (S | 1) +
          RO_ROWS * Aact + 2 * RO_ROWS * bufw + RO_ROWS);
}

extern "C" void launch_fused_rollout(
    float* state, float* step_ct, const float* w1, const float* b1,
    const float* w2, const float* b2, const float* w3, const float* b3,
    const float* Amat, const float* Bmat, const float* eps,
    const float* noise, float* st_obs, float* st_act, float* st_logp,
    float* st_nobs, float* st_rew, bool* st_done, int B, int S, int H1,
    int H2, int Aact, int T, float max_steps, float inv_softplus_bias,
    float scale_lb, void* stream) {
  const int blocks = (B + RO_ROWS - 1) / RO_ROWS;
  const int lds = fused_rollout_lds_bytes(S, H1, H2, Aact);
  hipLaunchKernelGGL(fused_rollout_kernel, dim3(blocks), dim3(RO_THREADS),
                     lds, (hipStream_t)stream, state, step_ct, w1, b1, w2, b2,
                     w3, b3, Amat, Bmat, eps, noise, st_obs, st_act, st_logp,
                     st_nobs, st_rew, st_done, B, S, H1, H2, Aact, T,
                     max_steps, inv_softplus_bias, scale_lb);
}


// ---------------------------------------------------------------------------
// MFMA rollout variant: the policy MLP on the matrix cores.
//
// Same loop/store/env semantics as fused_rollout_kernel, but the three
// policy GEMMs use v_mfma_f32_16x16x32_bf16 against the actor's bf16
// weight caches (the SAME caches the update phase computes with, so
// rollout-time log-probs and epoch-1 recomputed log-probs now agree to
// the kernel's fp32 reductions).  16 env rows per workgroup (the MFMA
// tile height); 4 waves split the H/16 output col tiles.  The env
// transition stays fp32 VALU (S ~ 17: not GEMM-shaped).
// Requirements (launcher-guarded): H1 == H2, H % 32 == 0, 2A <= 16.
// ---------------------------------------------------------------------------

namespace {

using ro_bfrag = __attribute__((ext_vector_type(8))) short;
using ro_ffrag = __attribute__((ext_vector_type(4))) float;

__device__ __forceinline__ float ro_fast_tanh(float x) {
  const float cx = fminf(fmaxf(x, -15.f), 15.f);
  const float t = __expf(2.f * cx);
  return (t - 1.f) / (t + 1.f);
}
__device__ __forceinline__ float ro_fast_softplus(float x) {
  if (x > 20.f) return x;
  return __logf(1.f + __expf(x));
}

__device__ __forceinline__ void ro_gemm_tile(
    const __hip_bfloat16* s_a, int lda, const __hip_bfloat16* s_w, int ldw,
    int j0, int Kp, int lane, ro_ffrag* acc) {
  const int row = lane & 15;
  const int koff = 8 * (lane >> 4);
  for (int kc = 0; kc < Kp; kc += 32) {
    const ro_bfrag a = *reinterpret_cast<const ro_bfrag*>(
        &s_a[(size_t)row * lda + kc + koff]);
    const ro_bfrag b = *reinterpret_cast<const ro_bfrag*>(
        &s_w[(size_t)(j0 + row) * ldw + kc + koff]);
    *acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, *acc, 0, 0, 0);
  }
}

__global__ void __launch_bounds__(RO_THREADS) fused_rollout_mfma_kernel(
    float* __restrict__ state, float* __restrict__ step_ct,
    const __hip_bfloat16* __restrict__ w1, const __hip_bfloat16* __restrict__ b1,
    const __hip_bfloat16* __restrict__ w2, const __hip_bfloat16* __restrict__ b2,
    const __hip_bfloat16* __restrict__ w3, const __hip_bfloat16* __restrict__ b3,
    const float* __restrict__ Amat, const float* __restrict__ Bmat,
    const float* __restrict__ eps, const float* __restrict__ noise,
    float* __restrict__ st_obs, float* __restrict__ st_act,
    float* __restrict__ st_logp, float* __restrict__ st_nobs,
    float* __restrict__ st_rew, bool* __restrict__ st_done, const int B,
    const int S, const int H, const int Aact, const int T,
    const float max_steps, const float inv_softplus_bias,
    const float scale_lb) {
  extern __shared__ char smem_raw[];
  const int Sp = (S + 31) & ~31;
  const int lx = Sp + 8, lh = H + 8;
  const int A2 = 2 * Aact;
  const int as = S | 1;
  char* ptr = smem_raw;
  auto alloc = [&](size_t bytes) {
    char* r = ptr;
    ptr += (bytes + 15) & ~size_t(15);
    return r;
  };
  __hip_bfloat16* s_w1 = (__hip_bfloat16*)alloc((size_t)H * lx * 2);
  __hip_bfloat16* s_w2 = (__hip_bfloat16*)alloc((size_t)H * lh * 2);
  __hip_bfloat16* s_w3 = (__hip_bfloat16*)alloc((size_t)16 * lh * 2);
  float* s_bias = (float*)alloc((size_t)(2 * H + 16) * 4);
  float* s_A = (float*)alloc((size_t)S * as * 4);
  float* s_B = (float*)alloc((size_t)Aact * as * 4);
  float* s_state = (float*)alloc((size_t)16 * S * 4);
  float* s_next = (float*)alloc((size_t)16 * S * 4);
  __hip_bfloat16* s_sbf = (__hip_bfloat16*)alloc((size_t)16 * lx * 2);
  __hip_bfloat16* s_h1 = (__hip_bfloat16*)alloc((size_t)16 * lh * 2);
  __hip_bfloat16* s_h2 = (__hip_bfloat16*)alloc((size_t)16 * lh * 2);
  float* s_head = (float*)alloc((size_t)16 * 16 * 4);
  float* s_act = (float*)alloc((size_t)16 * Aact * 4);
  float* s_lp = (float*)alloc((size_t)16 * Aact * 4);
  float* s_ct = (float*)alloc((size_t)16 * 4);

  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int row0 = blockIdx.x * 16;
  const int rows = min(16, B - row0);

  for (int i = tid; i < H * lx; i += RO_THREADS) {
    const int j = i / lx, k = i % lx;
    s_w1[i] = (k < S) ? w1[(size_t)j * S + k] : __hip_bfloat16(0.f);
  }
  for (int i = tid; i < H * lh; i += RO_THREADS) {
    const int j = i / lh, k = i % lh;
    s_w2[i] = (k < H) ? w2[(size_t)j * H + k] : __hip_bfloat16(0.f);
  }
  for (int i = tid; i < 16 * lh; i += RO_THREADS) {
    const int j = i / lh, k = i % lh;
    s_w3[i] = (j < A2 && k < H) ? w3[(size_t)j * H + k] : __hip_bfloat16(0.f);
  }
  for (int i = tid; i < H; i += RO_THREADS) {
    s_bias[i] = __bfloat162float(b1[i]);
    s_bias[H + i] = __bfloat162float(b2[i]);
  }
  for (int i = tid; i < A2; i += RO_THREADS)
    s_bias[2 * H + i] = __bfloat162float(b3[i]);
  for (int i = tid; i < S * S; i += RO_THREADS)
    s_A[(i / S) * as + i % S] = Amat[i];
  for (int i = tid; i < Aact * S; i += RO_THREADS)
    s_B[(i / S) * as + i % S] = Bmat[i];
  for (int i = tid; i < rows * S; i += RO_THREADS)
    s_state[i] = state[(size_t)(row0 + i / S) * S + i % S];
  for (int r = tid; r < rows; r += RO_THREADS) s_ct[r] = step_ct[row0 + r];
  __syncthreads();

  const float LOG_SQRT_2PI = 0.9189385332046727f;
  const float LOG2 = 0.6931471805599453f;
  const float lim = 1.0f - 1.1920929e-7f;
  const int erow = (lane >> 4) * 4;
  const int ecol = lane & 15;

  for (int t = 0; t < T; ++t) {
    // store obs + restage the bf16 policy input (zero pads)
    for (int i = tid; i < 16 * lx; i += RO_THREADS) {
      const int r = i / lx, k = i % lx;
      s_sbf[i] = (r < rows && k < S) ? __hip_bfloat16(s_state[r * S + k])
                                     : __hip_bfloat16(0.f);
    }
    for (int i = tid; i < rows * S; i += RO_THREADS)
      st_obs[((size_t)(row0 + i / S) * T + t) * S + i % S] =
          s_state[(i / S) * S + i % S];
    __syncthreads();
    // layer 1
    for (int ct = wave; ct < H / 16; ct += 4) {
      ro_ffrag acc = {};
      ro_gemm_tile(s_sbf, lx, s_w1, lx, ct * 16, Sp, lane, &acc);
      const int col = ct * 16 + ecol;
      const float bias = s_bias[col];
#pragma unroll
      for (int q = 0; q < 4; ++q)
        s_h1[(size_t)(erow + q) * lh + col] =
            __hip_bfloat16(ro_fast_tanh(acc[q] + bias));
    }
    __syncthreads();
    // layer 2
    for (int ct = wave; ct < H / 16; ct += 4) {
      ro_ffrag acc = {};
      ro_gemm_tile(s_h1, lh, s_w2, lh, ct * 16, H, lane, &acc);
      const int col = ct * 16 + ecol;
      const float bias = s_bias[H + col];
#pragma unroll
      for (int q = 0; q < 4; ++q)
        s_h2[(size_t)(erow + q) * lh + col] =
            __hip_bfloat16(ro_fast_tanh(acc[q] + bias));
    }
    __syncthreads();
    // heads (one 16x16 tile; wave 0)
    if (wave == 0) {
      ro_ffrag acc = {};
      ro_gemm_tile(s_h2, lh, s_w3, lh, 0, H, lane, &acc);
#pragma unroll
      for (int q = 0; q < 4; ++q)
        s_head[(size_t)(erow + q) * 16 + ecol] =
            acc[q] + (ecol < A2 ? s_bias[2 * H + ecol] : 0.f);
    }
    __syncthreads();
    // sample + per-element log-prob
    for (int i = tid; i < rows * Aact; i += RO_THREADS) {
      const int r = i / Aact, a = i % Aact;
      const float loc = s_head[r * 16 + a];
      float scale =
          ro_fast_softplus(s_head[r * 16 + Aact + a] + inv_softplus_bias);
      scale = fmaxf(scale, scale_lb);
      const float e = eps[((size_t)t * B + row0 + r) * Aact + a];
      const float u = loc + scale * e;
      float act = ro_fast_tanh(u);
      act = fminf(fmaxf(act, -lim), lim);
      s_act[r * Aact + a] = act;
      st_act[((size_t)(row0 + r) * T + t) * Aact + a] = act;
      s_lp[r * Aact + a] =
          -0.5f * e * e - __logf(scale) - LOG_SQRT_2PI
          - 2.0f * (LOG2 - u - ro_fast_softplus(-2.0f * u));
    }
    __syncthreads();
    for (int r = tid; r < rows; r += RO_THREADS) {
      float sum = 0.f;
      for (int a = 0; a < Aact; ++a) sum += s_lp[r * Aact + a];
      st_logp[(size_t)(row0 + r) * T + t] = sum;
    }
    // env transition (fp32, same math/order as the VALU kernel)
    for (int i = tid; i < rows * S; i += RO_THREADS) {
      const int r = i / S, j = i % S;
      float acc = 0.f;
      const float* sr = &s_state[r * S];
#pragma unroll 4
      for (int k = 0; k < S; ++k) acc += sr[k] * s_A[k * as + j];
      const float* ar = &s_act[r * Aact];
#pragma unroll
      for (int k = 0; k < Aact; ++k) acc += ar[k] * s_B[k * as + j];
      s_next[r * S + j] = tanhf(acc);
    }
    __syncthreads();
    for (int i = tid; i < rows * S; i += RO_THREADS) {
      const int r = i / S, j = i % S;
      const bool trunc = (s_ct[r] + 1.f) >= max_steps;
      const float ns = s_next[r * S + j];
      st_nobs[((size_t)(row0 + r) * T + t) * S + j] = ns;
      const float carry =
          trunc ? noise[((size_t)t * B + row0 + r) * S + j] : ns;
      s_next[r * S + j] = carry;  // committed after the sync below
      if (j == 0) {
        float ctrl = 0.f;
#pragma unroll
        for (int k = 0; k < Aact; ++k)
          ctrl += s_act[r * Aact + k] * s_act[r * Aact + k];
        st_rew[(size_t)(row0 + r) * T + t] = ns - 0.1f * ctrl;
        st_done[(size_t)(row0 + r) * T + t] = trunc;
      }
    }
    __syncthreads();
    for (int i = tid; i < rows * S; i += RO_THREADS)
      s_state[i] = s_next[i];
    for (int r = tid; r < rows; r += RO_THREADS) {
      const bool trunc = (s_ct[r] + 1.f) >= max_steps;
      s_ct[r] = trunc ? 0.f : s_ct[r] + 1.f;
    }
    __syncthreads();
  }

  for (int i = tid; i < rows * S; i += RO_THREADS)
    state[(size_t)(row0 + i / S) * S + i % S] = s_state[i];
  for (int r = tid; r < rows; r += RO_THREADS) step_ct[row0 + r] = s_ct[r];
}

}  // namespace

extern "C" int fused_rollout_mfma_ok(int S, int H1, int H2, int Aact) {
  if (H1 != H2 || H1 % 32 != 0 || 2 * Aact > 16) return 0;
  const int Sp = (S + 31) & ~31;
  const int lx = Sp + 8, lh = H1 + 8, as = S | 1;
  const size_t lds =
      (size_t)H1 * lx * 2 + (size_t)H1 * lh * 2 + 16 * lh * 2 +
      (2 * H1 + 16) * 4 + (size_t)S * as * 4 + (size_t)Aact * as * 4 +
      16 * S * 8 + 16 * lx * 2 + 2 * 16 * lh * 2 + 16 * 16 * 4 +
      16 * Aact * 8 + 64 + 16 * 16;  // + alignment slack
  return lds <= 160 * 1024;
}

extern "C" void launch_fused_rollout_mfma(
    float* state, float* step_ct, const void* w1, const void* b1,
    const void* w2, const void* b2, const void* w3, const void* b3,
    const float* Amat, const float* Bmat, const float* eps,
    const float* noise, float* st_obs, float* st_act, float* st_logp,
    float* st_nobs, float* st_rew, bool* st_done, int B, int S, int H1,
    int H2, int Aact, int T, float max_steps, float inv_softplus_bias,
    float scale_lb, void* stream) {
  const int blocks = (B + 15) / 16;
  const int Sp = (S + 31) & ~31;
  const int lx = Sp + 8, lh = H1 + 8, as = S | 1;
  const size_t lds =
      (size_t)H1 * lx * 2 + (size_t)H1 * lh * 2 + 16 * lh * 2 +
      (2 * H1 + 16) * 4 + (size_t)S * as * 4 + (size_t)Aact * as * 4 +
      16 * S * 8 + 16 * lx * 2 + 2 * 16 * lh * 2 + 16 * 16 * 4 +
      16 * Aact * 8 + 64 + 16 * 16;
  hipLaunchKernelGGL(fused_rollout_mfma_kernel, dim3(blocks),
                     dim3(RO_THREADS), (int)lds, (hipStream_t)stream, state,
                     step_ct, (const __hip_bfloat16*)w1,
                     (const __hip_bfloat16*)b1, (const __hip_bfloat16*)w2,
                     (const __hip_bfloat16*)b2, (const __hip_bfloat16*)w3,
                     (const __hip_bfloat16*)b3, Amat, Bmat, eps, noise,
                     st_obs, st_act, st_logp, st_nobs, st_rew, st_done, B, S,
                     H1, Aact, T, max_steps, inv_softplus_bias, scale_lb);
}
