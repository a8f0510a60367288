// Fused synthetic-MuJoCo env step (CDNA4, gfx950).
//
// The eager step of rl_amd/envs/custom/synthetic.py::_step is ~12
// kernel launches (two GEMMs, tanh, clamp, pow/sum, adds, compares,
// clone) of ~4 us each — 16-64 of these per hipGraph-captured rollout.
// This kernel does the whole transition in ONE launch:
//
//   a      = clamp(action, -1, 1)
//   s'     = tanh(s @ A + a @ B)        (state updated IN-PLACE)
//   obs    = s'                          (written to the output buffer)
//   reward = s'[0] - 0.1 * sum(a^2)
//   t     += 1;  done = truncated = (t >= max_steps)
//
// Design: A [S,S] and B [Aact,S] staged in LDS with padded rows (bank
// rules), 16 env rows per workgroup (B=4096 -> 256 WGs), outputs
// computed thread-per-(row, state-dim).
//
// Numerics validated against the eager torch step in tests/test_ops.py.

#include <hip/hip_runtime.h>

#define ENV_THREADS 256
#define ENV_ROWS 16

namespace {

__global__ void __launch_bounds__(ENV_THREADS) synthetic_env_step_kernel(
    float* __restrict__ state,        // [B, S] in/out
    const float* __restrict__ action, // [B, Aact]
    const float* __restrict__ Amat,   // [S, S] row-major (k, j)
    const float* __restrict__ Bmat,   // [Aact, S]
    float* __restrict__ t,            // [B, 1] in/out
    float* __restrict__ obs_out,      // next obs, row stride obs_rs
    float* __restrict__ prev_out,     // pre-step obs (nullable), row stride obs_rs
    float* __restrict__ reward,       // row stride sc_rs
    bool* __restrict__ done,          // row stride sc_rs
    const float* __restrict__ reset_noise,  // [B, S] (nullable): auto-reset
    const long obs_rs, const long sc_rs, const long act_rs,
    const int Bn, const int S, const int Aact, const float max_steps) {
  extern __shared__ float smem[];
  const int apad = S | 1;  // odd stride: no LDS bank collisions
  float* s_A = smem;                 // [S, apad]
  float* s_B = s_A + S * apad;       // [Aact, apad]
  float* s_state = s_B + Aact * apad;  // [ENV_ROWS, apad] (old state)
  float* s_act = s_state + ENV_ROWS * apad;  // [ENV_ROWS, Aact]
  float* s_t = s_act + ENV_ROWS * Aact;      // [ENV_ROWS] (old t)

  const int tid = threadIdx.x;
  const int row0 = blockIdx.x * ENV_ROWS;
  const int rows = min(ENV_ROWS, Bn - row0);

  for (int i = tid; i < S * S; i += ENV_THREADS)
    s_A[(i / S) * apad + i % S] = Amat[i];
  for (int i = tid; i < Aact * S; i += ENV_THREADS)
    s_B[(i / S) * apad + i % S] = Bmat[i];
  for (int i = tid; i < rows * S; i += ENV_THREADS) {
    const int r = i / S;
    s_state[r * apad + i % S] = state[(size_t)(row0 + r) * S + i % S];
  }
  for (int i = tid; i < rows * Aact; i += ENV_THREADS) {
    const int r = i / Aact;
    float a = action[(size_t)(row0 + r) * act_rs + i % Aact];
    s_act[r * Aact + i % Aact] = fminf(1.f, fmaxf(-1.f, a));
  }
  for (int r = tid; r < rows; r += ENV_THREADS) s_t[r] = t[row0 + r];
  __syncthreads();

  for (int i = tid; i < rows * S; i += ENV_THREADS) {
    const int r = i / S;
    const int j = i % S;
    float acc = 0.f;
    const float* sr = &s_state[r * apad];
#pragma unroll 4
    for (int k = 0; k < S; ++k) acc += sr[k] * s_A[k * apad + j];
    const float* ar = &s_act[r * Aact];
#pragma unroll
    for (int k = 0; k < Aact; ++k) acc += ar[k] * s_B[k * apad + j];
    const float ns = tanhf(acc);
    const int gr = row0 + r;
    const bool trunc = (s_t[r] + 1.f) >= max_steps;
    // step_and_maybe_reset semantics: the STORE sees the terminal obs,
    // the carried state auto-resets (noise) so the next step continues
    // a fresh trajectory with zero host involvement.
    float carry = ns;
    if (reset_noise != nullptr && trunc)
      carry = reset_noise[(size_t)gr * S + j];
    state[(size_t)gr * S + j] = carry;
    obs_out[(size_t)gr * obs_rs + j] = ns;
    if (prev_out != nullptr) prev_out[(size_t)gr * obs_rs + j] = sr[j];
    if (j == 0) {
      float ctrl = 0.f;
#pragma unroll
      for (int k = 0; k < Aact; ++k) ctrl += ar[k] * ar[k];
      t[gr] = (reset_noise != nullptr && trunc) ? 0.f : s_t[r] + 1.f;
      reward[(size_t)gr * sc_rs] = ns - 0.1f * ctrl;
      done[(size_t)gr * sc_rs] = trunc;
    }
  }
}

}  // namespace

extern "C" int synthetic_env_step_lds_bytes(int S, int Aact) {
  const int apad = S | 1;
  return (int)sizeof(float) *
         (S * apad + Aact * apad + ENV_ROWS * apad + ENV_ROWS * Aact +
          ENV_ROWS);
}

extern "C" void launch_synthetic_env_step(float* state, const float* action,
                                          const float* Amat, const float* Bmat,
                                          float* t, float* obs_out,
                                          float* prev_out, float* reward,
                                          bool* done, const float* reset_noise,
                                          long obs_rs, long sc_rs, long act_rs,
                                          int Bn, int S, int Aact,
                                          float max_steps, void* stream) {
  const int blocks = (Bn + ENV_ROWS - 1) / ENV_ROWS;
  const int lds = synthetic_env_step_lds_bytes(S, Aact);
  hipLaunchKernelGGL(synthetic_env_step_kernel, dim3(blocks),
                     dim3(ENV_THREADS), lds, (hipStream_t)stream, state,
                     action, Amat, Bmat, t, obs_out, prev_out, reward, done,
                     reset_noise, obs_rs, sc_rs, act_rs, Bn, S, Aact,
                     max_steps);
}
