// Fused 3-layer MLP forward + backward-dgrad for the PPO update phase
// (CDNA4, gfx950).
//
// The minibatch update runs actor and critic MLPs
// (Linear(O,H) -> tanh -> Linear(H,H) -> tanh -> Linear(H,A2)) under
// autograd: ~15 forward and ~30 backward elementwise/GEMM launches per
// network per minibatch at [16384, 64] shapes where launch overhead
// dominates.  Two kernels replace them:
//
//  * mlp3_fwd_kernel — whole forward in one launch (weights staged in
//    LDS with padded rows, 8 rows per workgroup like fused_actor.hip),
//    saving the tanh activations h1, h2 (bf16) for the backward.
//  * mlp3_bwd_kernel — the gradient chain
//        dh2 = (dOut @ W3) * (1 - h2^2)
//        dh1 = (dh2  @ W2) * (1 - h1^2)
//    in one launch (no dX: the MLP input is the rollout observation, a
//    leaf).  The three weight gradients then go through the MFMA
//    split-K wgrad (wgrad.hip) on (dOut, h2), (dh2, h1), (dh1, x).
//
// bf16 I/O and weights, fp32 accumulation and tanh, matching
// autocast's eager numerics.  Validated against the eager MLP
// fwd/bwd in tests/test_ops.py.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define MLP_THREADS 256
#define MLP_ROWS 8

namespace {

__device__ __forceinline__ float b2f(__hip_bfloat16 v) {
  return __bfloat162float(v);
}
__device__ __forceinline__ __hip_bfloat16 f2b(float v) {
  return __hip_bfloat16(v);
}

// LDS layout helper: weights [rows, cols] stored with stride cols+1
__global__ void __launch_bounds__(MLP_THREADS) mlp3_fwd_kernel(
    const __hip_bfloat16* __restrict__ x,   // [N, O]
    const __hip_bfloat16* __restrict__ w1,  // [H, O]
    const __hip_bfloat16* __restrict__ b1,  // [H]
    const __hip_bfloat16* __restrict__ w2,  // [H, H]
    const __hip_bfloat16* __restrict__ b2,  // [H]
    const __hip_bfloat16* __restrict__ w3,  // [A2, H]
    const __hip_bfloat16* __restrict__ b3,  // [A2]
    __hip_bfloat16* __restrict__ out,       // [N, A2]
    __hip_bfloat16* __restrict__ h1_out,    // [N, H]
    __hip_bfloat16* __restrict__ h2_out,    // [N, H]
    const int N, const int O, const int H, const int A2) {
  extern __shared__ __hip_bfloat16 smem[];
  const int w1s = O + 1, w2s = H + 1, w3s = H + 1;
  __hip_bfloat16* s_w1 = smem;
  __hip_bfloat16* s_b1 = s_w1 + H * w1s;
  __hip_bfloat16* s_w2 = s_b1 + H;
  __hip_bfloat16* s_b2 = s_w2 + H * w2s;
  __hip_bfloat16* s_w3 = s_b2 + H;
  __hip_bfloat16* s_b3 = s_w3 + A2 * w3s;
  const int bufw = max(max(O, H), A2);
  __hip_bfloat16* buf_a = s_b3 + A2;            // [MLP_ROWS, bufw]
  __hip_bfloat16* buf_b = buf_a + MLP_ROWS * bufw;

  const int tid = threadIdx.x;
  const int row0 = blockIdx.x * MLP_ROWS;
  const int rows = min(MLP_ROWS, N - row0);

  for (int i = tid; i < H * O; i += MLP_THREADS) s_w1[(i / O) * w1s + i % O] = w1[i];
  for (int i = tid; i < H; i += MLP_THREADS) s_b1[i] = b1[i];
  for (int i = tid; i < H * H; i += MLP_THREADS) s_w2[(i / H) * w2s + i % H] = w2[i];
  for (int i = tid; i < H; i += MLP_THREADS) s_b2[i] = b2[i];
  for (int i = tid; i < A2 * H; i += MLP_THREADS) s_w3[(i / H) * w3s + i % H] = w3[i];
  for (int i = tid; i < A2; i += MLP_THREADS) s_b3[i] = b3[i];
  for (int i = tid; i < rows * O; i += MLP_THREADS)
    buf_a[(i / O) * bufw + i % O] = x[(size_t)(row0 + i / O) * O + i % O];
  __syncthreads();

  // layer 1 + tanh
  for (int i = tid; i < rows * H; i += MLP_THREADS) {
    const int r = i / H, j = i % H;
    float acc = b2f(s_b1[j]);
    const __hip_bfloat16* in = &buf_a[r * bufw];
    const __hip_bfloat16* wr = &s_w1[j * w1s];
#pragma unroll 4
    for (int k = 0; k < O; ++k) acc += b2f(in[k]) * b2f(wr[k]);
    const __hip_bfloat16 h = f2b(tanhf(acc));
    buf_b[r * bufw + j] = h;
    h1_out[(size_t)(row0 + r) * H + j] = h;
  }
  __syncthreads();
  // layer 2 + tanh
  for (int i = tid; i < rows * H; i += MLP_THREADS) {
    const int r = i / H, j = i % H;
    float acc = b2f(s_b2[j]);
    const __hip_bfloat16* in = &buf_b[r * bufw];
    const __hip_bfloat16* wr = &s_w2[j * w2s];
#pragma unroll 8
    for (int k = 0; k < H; ++k) acc += b2f(in[k]) * b2f(wr[k]);
    const __hip_bfloat16 h = f2b(tanhf(acc));
    buf_a[r * bufw + j] = h;
    h2_out[(size_t)(row0 + r) * H + j] = h;
  }
  __syncthreads();
  // layer 3 (linear head)
  for (int i = tid; i < rows * A2; i += MLP_THREADS) {
    const int r = i / A2, j = i % A2;
    float acc = b2f(s_b3[j]);
    const __hip_bfloat16* in = &buf_a[r * bufw];
    const __hip_bfloat16* wr = &s_w3[j * w3s];
#pragma unroll 8
    for (int k = 0; k < H; ++k) acc += b2f(in[k]) * b2f(wr[k]);
    out[(size_t)(row0 + r) * A2 + j] = f2b(acc);
  }
}

// dgrad chain: dh2 = (dOut @ W3) * (1-h2^2); dh1 = (dh2 @ W2) * (1-h1^2)
// NB: contraction runs over the OUTPUT index of each weight (W^T), so
// the LDS staging keeps weight layout and the dot strides over rows.
__global__ void __launch_bounds__(MLP_THREADS) mlp3_bwd_kernel(
    const __hip_bfloat16* __restrict__ dout,  // [N, A2]
    const __hip_bfloat16* __restrict__ h1,    // [N, H]
    const __hip_bfloat16* __restrict__ h2,    // [N, H]
    const __hip_bfloat16* __restrict__ w2,    // [H, H]
    const __hip_bfloat16* __restrict__ w3,    // [A2, H]
    __hip_bfloat16* __restrict__ dh1_out,     // [N, H]
    __hip_bfloat16* __restrict__ dh2_out,     // [N, H]
    const int N, const int H, const int A2) {
  extern __shared__ __hip_bfloat16 smem[];
  // store W3^T [H, A2] and W2^T [H, H] so the dgrad dots are row-contiguous
  const int w3ts = A2 + 1, w2ts = H + 1;
  __hip_bfloat16* s_w3t = smem;                 // [H, w3ts]
  __hip_bfloat16* s_w2t = s_w3t + H * w3ts;     // [H, w2ts]
  const int bufw = max(H, A2);
  __hip_bfloat16* buf_a = s_w2t + H * w2ts;     // [MLP_ROWS, bufw]
  __hip_bfloat16* buf_b = buf_a + MLP_ROWS * bufw;

  const int tid = threadIdx.x;
  const int row0 = blockIdx.x * MLP_ROWS;
  const int rows = min(MLP_ROWS, N - row0);

  for (int i = tid; i < A2 * H; i += MLP_THREADS) {
    const int r = i / H, c = i % H;  // w3[r][c] -> w3t[c][r]
    s_w3t[c * w3ts + r] = w3[i];
  }
  for (int i = tid; i < H * H; i += MLP_THREADS) {
    const int r = i / H, c = i % H;
    s_w2t[c * w2ts + r] = w2[i];
  }
  for (int i = tid; i < rows * A2; i += MLP_THREADS)
    buf_a[(i / A2) * bufw + i % A2] = dout[(size_t)(row0 + i / A2) * A2 + i % A2];
  __syncthreads();

  // dh2[j] = sum_a dout[a] * w3[a][j] = dot(dout_row, w3t[j]); * (1-h2^2)
  for (int i = tid; i < rows * H; i += MLP_THREADS) {
    const int r = i / H, j = i % H;
    float acc = 0.f;
    const __hip_bfloat16* in = &buf_a[r * bufw];
    const __hip_bfloat16* wr = &s_w3t[j * w3ts];
#pragma unroll 4
    for (int k = 0; k < A2; ++k) acc += b2f(in[k]) * b2f(wr[k]);
    const float hv = b2f(h2[(size_t)(row0 + r) * H + j]);
    const __hip_bfloat16 g = f2b(acc * (1.f - hv * hv));
    buf_b[r * bufw + j] = g;
    dh2_out[(size_t)(row0 + r) * H + j] = g;
  }
  __syncthreads();
  // dh1[j] = dot(dh2_row, w2t[j]) * (1-h1^2)
  for (int i = tid; i < rows * H; i += MLP_THREADS) {
    const int r = i / H, j = i % H;
    float acc = 0.f;
    const __hip_bfloat16* in = &buf_b[r * bufw];
    const __hip_bfloat16* wr = &s_w2t[j * w2ts];
#pragma unroll 8
    for (int k = 0; k < H; ++k) acc += b2f(in[k]) * b2f(wr[k]);
    const float hv = b2f(h1[(size_t)(row0 + r) * H + j]);
    dh1_out[(size_t)(row0 + r) * H + j] = f2b(acc * (1.f - hv * hv));
  }
}

}  // namespace

extern "C" int mlp3_lds_bytes(int O, int H, int A2) {
  const int bufw = max(max(O, H), A2);
  const int fwd = (int)sizeof(__hip_bfloat16) *
                  (H * (O + 1) + H + H * (H + 1) + H + A2 * (H + 1) + A2 +
                   2 * MLP_ROWS * bufw);
  const int bwd = (int)sizeof(__hip_bfloat16) *
                  (H * (A2 + 1) + H * (H + 1) + 2 * MLP_ROWS * bufw);
  return fwd > bwd ? fwd : bwd;
}

extern "C" void launch_mlp3_fwd(const void* x, const void* w1, const void* b1,
                                const void* w2, const void* b2, const void* w3,
                                const void* b3, void* out, void* h1, void* h2,
                                int N, int O, int H, int A2, void* stream) {
  const int blocks = (N + MLP_ROWS - 1) / MLP_ROWS;
  const int bufw = max(max(O, H), A2);
  const int lds = (int)sizeof(__hip_bfloat16) *
                  (H * (O + 1) + H + H * (H + 1) + H + A2 * (H + 1) + A2 +
                   2 * MLP_ROWS * bufw);
  hipLaunchKernelGGL(mlp3_fwd_kernel, dim3(blocks), dim3(MLP_THREADS), lds,
                     (hipStream_t)stream, (const __hip_bfloat16*)x,
                     (const __hip_bfloat16*)w1, (const __hip_bfloat16*)b1,
                     (const __hip_bfloat16*)w2, (const __hip_bfloat16*)b2,
                     (const __hip_bfloat16*)w3, (const __hip_bfloat16*)b3,
                     (__hip_bfloat16*)out, (__hip_bfloat16*)h1,
                     (__hip_bfloat16*)h2, N, O, H, A2);
}

extern "C" void launch_mlp3_bwd(const void* dout, const void* h1,
                                const void* h2, const void* w2, const void* w3,
                                void* dh1, void* dh2, int N, int H, int A2,
                                void* stream) {
  const int blocks = (N + MLP_ROWS - 1) / MLP_ROWS;
  const int bufw = max(H, A2);
  const int lds = (int)sizeof(__hip_bfloat16) *
                  (H * (A2 + 1) + H * (H + 1) + 2 * MLP_ROWS * bufw);
  hipLaunchKernelGGL(mlp3_bwd_kernel, dim3(blocks), dim3(MLP_THREADS), lds,
                     (hipStream_t)stream, (const __hip_bfloat16*)dout,
                     (const __hip_bfloat16*)h1, (const __hip_bfloat16*)h2,
                     (const __hip_bfloat16*)w2, (const __hip_bfloat16*)w3,
                     (__hip_bfloat16*)dh1, (__hip_bfloat16*)dh2, N, H, A2);
}

// ---------------------------------------------------------------------------
// MFMA whole-MLP kernels (v2).  The VALU kernels above win only for
// launch-bound small batches; at the PPO update's 16K-row minibatches
// their dot loops lose to hipBLASLt (measured r37: T=64 7.49 ms vs
// 5.18).  These variants put the three GEMMs on the matrix cores
// (v_mfma_f32_16x16x32_bf16, fragment mapping hardware-verified in
// benchmarks/mfma_probe.hip) with the whole layer chain fused:
//
//  * mlp3_mfma_fwd_kernel — 64 rows per workgroup (grid = N/64: 256+
//    workgroups at bench sizes — fills the 8 XCDs), weights staged in
//    LDS once.  W1/W2/W3 are [out,in] row-major, which IS the B^T
//    fragment layout: b[reg] = W[j][k0+reg] is a single 16-byte LDS
//    read per fragment (rows padded to 16-byte alignment).  tanh +
//    bias run in registers; h1/h2 (bf16) go to LDS for the next layer
//    and to global for the backward; fp32 input is converted during
//    staging (kills the eager obs->bf16 cast kernel).
//  * mlp3_mfma_bwd_kernel — the dgrad chain in one launch; W2/W3 are
//    staged TRANSPOSED so dX = dY @ W also reads single-b128 B
//    fragments; tanh' multiplies against the saved activations.
//
// Eager semantics matched: bf16 inputs/weights, fp32 accumulation,
// fp32 tanh, bf16 activations (= autocast's eager rounding).
// ---------------------------------------------------------------------------

namespace {

using mfrag_b = __attribute__((ext_vector_type(8))) short;
using mfrag_f = __attribute__((ext_vector_type(4))) float;

// Rows per workgroup is a template parameter: 32 (2 row blocks x 16,
// waves split row-block x col-parity) for launches that would not fill
// the chip at 64 — a 64-row tile at 16K rows is 256 WGs = 1 WG/CU with
// no latency hiding (measured 20 us/fwd); 64 (4 waves x 16 rows) once
// N/64 >= 512 WGs, where the halved per-WG weight staging wins
// (measured: 32-row tile cost ~5% at 65K rows).
#define M3_ROWS_SWITCH 32768

__device__ __forceinline__ int m3_pad32(int k) { return (k + 31) & ~31; }

// one 16x16 output tile over the full K: A rows from s_a (stride lda,
// 16B-aligned), B rows = weight rows j (stride ldb) — both b128 reads.
__device__ __forceinline__ void m3_gemm_tile(
    const __hip_bfloat16* s_a, int lda, const __hip_bfloat16* s_w, int ldb,
    int j0, int Kp, int lane, mfrag_f* acc) {
  const int row = lane & 15;
  const int koff = 8 * (lane >> 4);
  for (int kc = 0; kc < Kp; kc += 32) {
    const mfrag_b a = *reinterpret_cast<const mfrag_b*>(
        &s_a[(size_t)row * lda + kc + koff]);
    const mfrag_b b = *reinterpret_cast<const mfrag_b*>(
        &s_w[(size_t)(j0 + row) * ldb + kc + koff]);
    *acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, *acc, 0, 0, 0);
  }
}

template <typename TX, int R>
__device__ void mlp3_mfma_fwd_impl(
    const TX* __restrict__ x,               // [N, O]
    const __hip_bfloat16* __restrict__ w1,  // [H, O]
    const __hip_bfloat16* __restrict__ b1,  // [H]
    const __hip_bfloat16* __restrict__ w2,  // [H, H]
    const __hip_bfloat16* __restrict__ b2,  // [H]
    const __hip_bfloat16* __restrict__ w3,  // [A2, H]
    const __hip_bfloat16* __restrict__ b3,  // [A2]
    __hip_bfloat16* __restrict__ out,       // [N, A2]
    __hip_bfloat16* __restrict__ h1_out,    // [N, H]
    __hip_bfloat16* __restrict__ h2_out,    // [N, H]
    __hip_bfloat16* __restrict__ xb_out,    // [N, O] bf16 copy (wgrad)
    const int N, const int O, const int H, const int A2) {
  extern __shared__ __hip_bfloat16 smem[];
  const int Op = m3_pad32(O), Hp = m3_pad32(H);
  const int lx = Op + 8, lh = Hp + 8;
  const int A2p = (A2 + 15) & ~15;
  __hip_bfloat16* s_x = smem;                  // [64][lx]
  __hip_bfloat16* s_w1 = s_x + R * lx;   // [H][lx]
  __hip_bfloat16* s_h1 = s_w1 + H * lx;        // [64][lh]
  __hip_bfloat16* s_w2 = s_h1 + R * lh;  // [H][lh]
  __hip_bfloat16* s_h2 = s_w2 + H * lh;        // [64][lh]
  __hip_bfloat16* s_w3 = s_h2 + R * lh;  // [A2p][lh]

  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const long row0 = (long)blockIdx.x * R;
  const int rows = (int)min((long)R, (long)N - row0);

  for (int i = tid; i < R * lx; i += MLP_THREADS) {
    const int r = i / lx, k = i % lx;
    __hip_bfloat16 v = __hip_bfloat16(0.f);
    if (r < rows && k < O) v = __hip_bfloat16((float)x[(row0 + r) * O + k]);
    s_x[i] = v;
    if (xb_out != nullptr && r < rows && k < O)
      xb_out[(row0 + r) * O + k] = v;
  }
  for (int i = tid; i < H * lx; i += MLP_THREADS) {
    const int j = i / lx, k = i % lx;
    s_w1[i] = (k < O) ? w1[(size_t)j * O + k] : __hip_bfloat16(0.f);
  }
  for (int i = tid; i < H * lh; i += MLP_THREADS) {
    const int j = i / lh, k = i % lh;
    s_w2[i] = (k < H) ? w2[(size_t)j * H + k] : __hip_bfloat16(0.f);
  }
  for (int i = tid; i < A2p * lh; i += MLP_THREADS) {
    const int j = i / lh, k = i % lh;
    s_w3[i] = (j < A2 && k < H) ? w3[(size_t)j * H + k] : __hip_bfloat16(0.f);
  }
  __syncthreads();

  const int erow = (lane >> 4) * 4;  // epilogue row base (+reg)
  const int ecol = lane & 15;
  const int rblk = (R == 64 ? wave : (R == 32 ? (wave & 1) : 0)) * 16;
  const int ct0 = (R == 64 ? 0 : (R == 32 ? (wave >> 1) : wave));
  const int cts = (R == 64 ? 1 : (R == 32 ? 2 : 4));
  // layer 1 + tanh (global h1 goes out later via a coalesced LDS copy)
  for (int ct = ct0; ct < H / 16; ct += cts) {
    mfrag_f acc = {};
    m3_gemm_tile(&s_x[(size_t)rblk * lx], lx, s_w1, lx, ct * 16, Op,
                 lane, &acc);
    const int col = ct * 16 + ecol;
    const float bias = __bfloat162float(b1[col]);
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = rblk + erow + r;
      s_h1[(size_t)row * lh + col] = __hip_bfloat16(tanhf(acc[r] + bias));
    }
  }
  // zero the k-pad of h tiles once (Hp > H only when H % 32 == 16)
  for (int i = tid; i < R; i += MLP_THREADS)
    for (int k = H; k < Hp; ++k) {
      s_h1[(size_t)i * lh + k] = __hip_bfloat16(0.f);
      s_h2[(size_t)i * lh + k] = __hip_bfloat16(0.f);
    }
  __syncthreads();
  // coalesced vectorized h1 store (8 bf16 per thread) + layer 2
  const int h8 = H / 8;  // H % 16 == 0
  for (int i = tid; i < rows * h8; i += MLP_THREADS) {
    const int r = i / h8, c8 = i - r * h8;
    *reinterpret_cast<uint4*>(&h1_out[(row0 + r) * H + 8 * c8]) =
        *reinterpret_cast<const uint4*>(&s_h1[(size_t)r * lh + 8 * c8]);
  }
  for (int ct = ct0; ct < H / 16; ct += cts) {
    mfrag_f acc = {};
    m3_gemm_tile(&s_h1[(size_t)rblk * lh], lh, s_w2, lh, ct * 16, Hp,
                 lane, &acc);
    const int col = ct * 16 + ecol;
    const float bias = __bfloat162float(b2[col]);
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = rblk + erow + r;
      s_h2[(size_t)row * lh + col] = __hip_bfloat16(tanhf(acc[r] + bias));
    }
  }
  __syncthreads();
  // coalesced h2 store + layer 3 (head result staged over s_h1)
  for (int i = tid; i < rows * h8; i += MLP_THREADS) {
    const int r = i / h8, c8 = i - r * h8;
    *reinterpret_cast<uint4*>(&h2_out[(row0 + r) * H + 8 * c8]) =
        *reinterpret_cast<const uint4*>(&s_h2[(size_t)r * lh + 8 * c8]);
  }
  for (int ct = ct0; ct < A2p / 16; ct += cts) {
    mfrag_f acc = {};
    m3_gemm_tile(&s_h2[(size_t)rblk * lh], lh, s_w3, lh, ct * 16, Hp,
                 lane, &acc);
    const int col = ct * 16 + ecol;
    const float bias = (col < A2) ? __bfloat162float(b3[col]) : 0.f;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = rblk + erow + r;
      s_h1[(size_t)row * lh + col] = __hip_bfloat16(acc[r] + bias);
    }
  }
  __syncthreads();
  for (int i = tid; i < rows * A2; i += MLP_THREADS) {
    const int r = i / A2, c = i - r * A2;
    out[(row0 + r) * A2 + c] = s_h1[(size_t)r * lh + c];
  }
}

template <typename TX, int R>
__global__ void __launch_bounds__(MLP_THREADS) mlp3_mfma_fwd_kernel(
    const TX* __restrict__ x, const __hip_bfloat16* __restrict__ w1,
    const __hip_bfloat16* __restrict__ b1,
    const __hip_bfloat16* __restrict__ w2,
    const __hip_bfloat16* __restrict__ b2,
    const __hip_bfloat16* __restrict__ w3,
    const __hip_bfloat16* __restrict__ b3, __hip_bfloat16* __restrict__ out,
    __hip_bfloat16* __restrict__ h1_out, __hip_bfloat16* __restrict__ h2_out,
    __hip_bfloat16* __restrict__ xb_out, const int N, const int O,
    const int H, const int A2) {
  mlp3_mfma_fwd_impl<TX, R>(x, w1, b1, w2, b2, w3, b3, out, h1_out, h2_out,
                            xb_out, N, O, H, A2);
}

// Dual-network forward: blockIdx.y selects the net (0 = actor, 1 =
// critic).  Both read the SAME input rows; the bf16 input copy for the
// wgrad is written once (net 0).
struct AC2Fwd {
  const __hip_bfloat16* w1[2];
  const __hip_bfloat16* b1[2];
  const __hip_bfloat16* w2[2];
  const __hip_bfloat16* b2[2];
  const __hip_bfloat16* w3[2];
  const __hip_bfloat16* b3[2];
  __hip_bfloat16* out[2];
  __hip_bfloat16* h1[2];
  __hip_bfloat16* h2[2];
  int H[2];
  int A2[2];
};

template <typename TX, int R>
__global__ void __launch_bounds__(MLP_THREADS) mlp3_mfma_fwd2_kernel(
    const TX* __restrict__ x, const AC2Fwd args,
    __hip_bfloat16* __restrict__ xb_out, const int N, const int O) {
  const int net = blockIdx.y;
  mlp3_mfma_fwd_impl<TX, R>(x, args.w1[net], args.b1[net], args.w2[net],
                            args.b2[net], args.w3[net], args.b3[net],
                            args.out[net], args.h1[net], args.h2[net],
                            net == 0 ? xb_out : nullptr, N, O, args.H[net],
                            args.A2[net]);
}

template <typename TD, int R>
__device__ void mlp3_mfma_bwd_impl(
    const TD* __restrict__ dout,            // [N, A2]
    const __hip_bfloat16* __restrict__ h1,  // [N, H]
    const __hip_bfloat16* __restrict__ h2,  // [N, H]
    const __hip_bfloat16* __restrict__ w2,  // [H, H]
    const __hip_bfloat16* __restrict__ w3,  // [A2, H]
    __hip_bfloat16* __restrict__ dh1_out,   // [N, H]
    __hip_bfloat16* __restrict__ dh2_out,   // [N, H]
    const int N, const int H, const int A2) {
  extern __shared__ __hip_bfloat16 smem[];
  const int A2p32 = m3_pad32(A2), Hp = m3_pad32(H);
  const int ld = A2p32 + 8, lh = Hp + 8;
  __hip_bfloat16* s_dy = smem;                   // [R][ld]
  __hip_bfloat16* s_w3t = s_dy + R * ld;   // [H][ld]  w3t[j][k]=w3[k][j]
  __hip_bfloat16* s_dh2 = s_w3t + H * ld;        // [R][lh]
  __hip_bfloat16* s_w2t = s_dh2 + R * lh;  // [H][lh]
  __hip_bfloat16* s_dh1 = s_w2t + H * lh;        // [R][lh]

  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const long row0 = (long)blockIdx.x * R;
  const int rows = (int)min((long)R, (long)N - row0);

  for (int i = tid; i < R * ld; i += MLP_THREADS) {
    const int r = i / ld, k = i % ld;
    s_dy[i] = (r < rows && k < A2)
                  ? __hip_bfloat16((float)dout[(row0 + r) * A2 + k])
                  : __hip_bfloat16(0.f);
  }
  for (int i = tid; i < H * ld; i += MLP_THREADS) {
    const int j = i / ld, k = i % ld;
    s_w3t[i] = (k < A2) ? w3[(size_t)k * H + j] : __hip_bfloat16(0.f);
  }
  for (int i = tid; i < H * lh; i += MLP_THREADS) {
    const int j = i / lh, k = i % lh;
    s_w2t[i] = (k < H) ? w2[(size_t)k * H + j] : __hip_bfloat16(0.f);
  }
  __syncthreads();

  const int erow = (lane >> 4) * 4;
  const int ecol = lane & 15;
  const int rblk = (R == 64 ? wave : (R == 32 ? (wave & 1) : 0)) * 16;
  const int ct0 = (R == 64 ? 0 : (R == 32 ? (wave >> 1) : wave));
  const int cts = (R == 64 ? 1 : (R == 32 ? 2 : 4));
  // dh2 = (dY @ W3) * (1 - h2^2)
  for (int ct = ct0; ct < H / 16; ct += cts) {
    mfrag_f acc = {};
    m3_gemm_tile(&s_dy[(size_t)rblk * ld], ld, s_w3t, ld, ct * 16,
                 A2p32, lane, &acc);
    const int col = ct * 16 + ecol;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = rblk + erow + r;
      float g = 0.f;
      if (row < rows) {
        const float hv = __bfloat162float(h2[(row0 + row) * H + col]);
        g = acc[r] * (1.f - hv * hv);
      }
      s_dh2[(size_t)row * lh + col] = __hip_bfloat16(g);
    }
  }
  for (int i = tid; i < R; i += MLP_THREADS)
    for (int k = H; k < Hp; ++k) s_dh2[(size_t)i * lh + k] = __hip_bfloat16(0.f);
  __syncthreads();
  // coalesced dh2 store + dh1 = (dh2 @ W2) * (1 - h1^2)
  const int h8 = H / 8;
  for (int i = tid; i < rows * h8; i += MLP_THREADS) {
    const int r = i / h8, c8 = i - r * h8;
    *reinterpret_cast<uint4*>(&dh2_out[(row0 + r) * H + 8 * c8]) =
        *reinterpret_cast<const uint4*>(&s_dh2[(size_t)r * lh + 8 * c8]);
  }
  for (int ct = ct0; ct < H / 16; ct += cts) {
    mfrag_f acc = {};
    m3_gemm_tile(&s_dh2[(size_t)rblk * lh], lh, s_w2t, lh, ct * 16, Hp,
                 lane, &acc);
    const int col = ct * 16 + ecol;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = rblk + erow + r;
      float g = 0.f;
      if (row < rows) {
        const float hv = __bfloat162float(h1[(row0 + row) * H + col]);
        g = acc[r] * (1.f - hv * hv);
      }
      s_dh1[(size_t)row * lh + col] = __hip_bfloat16(g);
    }
  }
  __syncthreads();
  for (int i = tid; i < rows * h8; i += MLP_THREADS) {
    const int r = i / h8, c8 = i - r * h8;
    *reinterpret_cast<uint4*>(&dh1_out[(row0 + r) * H + 8 * c8]) =
        *reinterpret_cast<const uint4*>(&s_dh1[(size_t)r * lh + 8 * c8]);
  }
}

template <typename TD, int R>
__global__ void __launch_bounds__(MLP_THREADS) mlp3_mfma_bwd_kernel(
    const TD* __restrict__ dout, const __hip_bfloat16* __restrict__ h1,
    const __hip_bfloat16* __restrict__ h2,
    const __hip_bfloat16* __restrict__ w2,
    const __hip_bfloat16* __restrict__ w3, __hip_bfloat16* __restrict__ dh1,
    __hip_bfloat16* __restrict__ dh2, const int N, const int H,
    const int A2) {
  mlp3_mfma_bwd_impl<TD, R>(dout, h1, h2, w2, w3, dh1, dh2, N, H, A2);
}

struct AC2Bwd {
  const __hip_bfloat16* dout[2];
  const __hip_bfloat16* h1[2];
  const __hip_bfloat16* h2[2];
  const __hip_bfloat16* w2[2];
  const __hip_bfloat16* w3[2];
  __hip_bfloat16* dh1[2];
  __hip_bfloat16* dh2[2];
  int H[2];
  int A2[2];
};

template <int R>
__global__ void __launch_bounds__(MLP_THREADS) mlp3_mfma_bwd2_kernel(
    const AC2Bwd args, const int N) {
  const int net = blockIdx.y;
  mlp3_mfma_bwd_impl<__hip_bfloat16, R>(
      args.dout[net], args.h1[net], args.h2[net], args.w2[net], args.w3[net],
      args.dh1[net], args.dh2[net], N, args.H[net], args.A2[net]);
}

}  // namespace

extern "C" int mlp3_mfma_lds_bytes(int O, int H, int A2) {
  const int R = 64;  // LDS bound checked for the larger tile
  const int Op = (O + 31) & ~31, Hp = (H + 31) & ~31;
  const int lx = Op + 8, lh = Hp + 8, A2p = (A2 + 15) & ~15;
  const int ld = ((A2 + 31) & ~31) + 8;
  const int fwd = 2 * (R * lx + H * lx + 2 * R * lh + H * lh + A2p * lh);
  const int bwd = 2 * (R * ld + H * ld + 2 * R * lh + H * lh);
  return fwd > bwd ? fwd : bwd;
}

extern "C" void launch_mlp3_mfma_fwd(const void* x, int x_is_f32,
                                     const void* w1, const void* b1,
                                     const void* w2, const void* b2,
                                     const void* w3, const void* b3,
                                     void* out, void* h1, void* h2, void* xb,
                                     long N, int O, int H, int A2,
                                     void* stream) {
  const int R = N >= M3_ROWS_SWITCH ? 64 : 32;
  const int blocks = (int)((N + R - 1) / R);
  const int Op = (O + 31) & ~31, Hp = (H + 31) & ~31;
  const int lx = Op + 8, lh = Hp + 8, A2p = (A2 + 15) & ~15;
  const int lds = 2 * (R * lx + H * lx + 2 * R * lh + H * lh + A2p * lh);
#define M3_LAUNCH_FWD(TX, RR)                                               \
  hipLaunchKernelGGL((mlp3_mfma_fwd_kernel<TX, RR>), dim3(blocks),          \
                     dim3(MLP_THREADS), lds, (hipStream_t)stream,           \
                     (const TX*)x, (const __hip_bfloat16*)w1,               \
                     (const __hip_bfloat16*)b1, (const __hip_bfloat16*)w2,  \
                     (const __hip_bfloat16*)b2, (const __hip_bfloat16*)w3,  \
                     (const __hip_bfloat16*)b3, (__hip_bfloat16*)out,       \
                     (__hip_bfloat16*)h1, (__hip_bfloat16*)h2,              \
                     (__hip_bfloat16*)xb, (int)N, O, H, A2)
  if (x_is_f32) {
    if (R == 64) M3_LAUNCH_FWD(float, 64);
    else M3_LAUNCH_FWD(float, 32);
  } else {
    if (R == 64) M3_LAUNCH_FWD(__hip_bfloat16, 64);
    else M3_LAUNCH_FWD(__hip_bfloat16, 32);
  }
#undef M3_LAUNCH_FWD
}

extern "C" void launch_mlp3_mfma_bwd(const void* dout, int d_is_f32,
                                     const void* h1, const void* h2,
                                     const void* w2, const void* w3,
                                     void* dh1, void* dh2, long N, int H,
                                     int A2, void* stream) {
  const int R = N >= M3_ROWS_SWITCH ? 64 : 32;
  const int blocks = (int)((N + R - 1) / R);
  const int ld = ((A2 + 31) & ~31) + 8, lh = ((H + 31) & ~31) + 8;
  const int lds = 2 * (R * ld + H * ld + 2 * R * lh + H * lh);
#define M3_LAUNCH_BWD(TD, RR)                                               \
  hipLaunchKernelGGL((mlp3_mfma_bwd_kernel<TD, RR>), dim3(blocks),          \
                     dim3(MLP_THREADS), lds, (hipStream_t)stream,           \
                     (const TD*)dout, (const __hip_bfloat16*)h1,            \
                     (const __hip_bfloat16*)h2, (const __hip_bfloat16*)w2,  \
                     (const __hip_bfloat16*)w3, (__hip_bfloat16*)dh1,       \
                     (__hip_bfloat16*)dh2, (int)N, H, A2)
  if (d_is_f32) {
    if (R == 64) M3_LAUNCH_BWD(float, 64);
    else M3_LAUNCH_BWD(float, 32);
  } else {
    if (R == 64) M3_LAUNCH_BWD(__hip_bfloat16, 64);
    else M3_LAUNCH_BWD(__hip_bfloat16, 32);
  }
#undef M3_LAUNCH_BWD
}

extern "C" void launch_mlp3_mfma_fwd2(const void* x, int x_is_f32,
                                      const void* const* w,  // 12 ptrs
                                      void* const* o,        // 6 ptrs
                                      void* xb, long N, int O, const int* H,
                                      const int* A2, void* stream) {
  AC2Fwd a;
  for (int n = 0; n < 2; ++n) {
    a.w1[n] = (const __hip_bfloat16*)w[n * 6 + 0];
    a.b1[n] = (const __hip_bfloat16*)w[n * 6 + 1];
    a.w2[n] = (const __hip_bfloat16*)w[n * 6 + 2];
    a.b2[n] = (const __hip_bfloat16*)w[n * 6 + 3];
    a.w3[n] = (const __hip_bfloat16*)w[n * 6 + 4];
    a.b3[n] = (const __hip_bfloat16*)w[n * 6 + 5];
    a.out[n] = (__hip_bfloat16*)o[n * 3 + 0];
    a.h1[n] = (__hip_bfloat16*)o[n * 3 + 1];
    a.h2[n] = (__hip_bfloat16*)o[n * 3 + 2];
    a.H[n] = H[n];
    a.A2[n] = A2[n];
  }
  const int R = N >= M3_ROWS_SWITCH ? 64 : 32;
  const int blocks = (int)((N + R - 1) / R);
  int lds = 0;
  for (int n = 0; n < 2; ++n) {
    const int Op = (O + 31) & ~31, Hp = (H[n] + 31) & ~31;
    const int lx = Op + 8, lh = Hp + 8, A2p = (A2[n] + 15) & ~15;
    const int b = 2 * (R * lx + H[n] * lx + 2 * R * lh + H[n] * lh + A2p * lh);
    if (b > lds) lds = b;
  }
  dim3 grid(blocks, 2);
  if (x_is_f32) {
    if (R == 64)
      hipLaunchKernelGGL((mlp3_mfma_fwd2_kernel<float, 64>), grid,
                         dim3(MLP_THREADS), lds, (hipStream_t)stream,
                         (const float*)x, a, (__hip_bfloat16*)xb, (int)N, O);
    else
      hipLaunchKernelGGL((mlp3_mfma_fwd2_kernel<float, 32>), grid,
                         dim3(MLP_THREADS), lds, (hipStream_t)stream,
                         (const float*)x, a, (__hip_bfloat16*)xb, (int)N, O);
  } else {
    if (R == 64)
      hipLaunchKernelGGL((mlp3_mfma_fwd2_kernel<__hip_bfloat16, 64>), grid,
                         dim3(MLP_THREADS), lds, (hipStream_t)stream,
                         (const __hip_bfloat16*)x, a, (__hip_bfloat16*)xb,
                         (int)N, O);
    else
      hipLaunchKernelGGL((mlp3_mfma_fwd2_kernel<__hip_bfloat16, 32>), grid,
                         dim3(MLP_THREADS), lds, (hipStream_t)stream,
                         (const __hip_bfloat16*)x, a, (__hip_bfloat16*)xb,
                         (int)N, O);
  }
}

extern "C" void launch_mlp3_mfma_bwd2(const void* const* dptr,  // 10 ptrs
                                      void* const* dh,          // 4 ptrs
                                      long N, const int* H, const int* A2,
                                      void* stream) {
  AC2Bwd a;
  for (int n = 0; n < 2; ++n) {
    a.dout[n] = (const __hip_bfloat16*)dptr[n * 5 + 0];
    a.h1[n] = (const __hip_bfloat16*)dptr[n * 5 + 1];
    a.h2[n] = (const __hip_bfloat16*)dptr[n * 5 + 2];
    a.w2[n] = (const __hip_bfloat16*)dptr[n * 5 + 3];
    a.w3[n] = (const __hip_bfloat16*)dptr[n * 5 + 4];
    a.dh1[n] = (__hip_bfloat16*)dh[n * 2 + 0];
    a.dh2[n] = (__hip_bfloat16*)dh[n * 2 + 1];
    a.H[n] = H[n];
    a.A2[n] = A2[n];
  }
  const int R = N >= M3_ROWS_SWITCH ? 64 : 32;
  const int blocks = (int)((N + R - 1) / R);
  int lds = 0;
  for (int n = 0; n < 2; ++n) {
    const int ld = ((A2[n] + 31) & ~31) + 8, lh = ((H[n] + 31) & ~31) + 8;
    const int b = 2 * (R * ld + H[n] * ld + 2 * R * lh + H[n] * lh);
    if (b > lds) lds = b;
  }
  dim3 grid(blocks, 2);
  if (R == 64)
    hipLaunchKernelGGL((mlp3_mfma_bwd2_kernel<64>), grid, dim3(MLP_THREADS),
                       lds, (hipStream_t)stream, a, (int)N);
  else
    hipLaunchKernelGGL((mlp3_mfma_bwd2_kernel<32>), grid, dim3(MLP_THREADS),
                       lds, (hipStream_t)stream, a, (int)N);
}

// Same-network PAIR forward (no gradients): grid.y picks which of two
// inputs to run — GAE's value / next_value evals in ONE launch.
template <typename TX, int R>
__global__ void __launch_bounds__(MLP_THREADS) mlp3_mfma_fwdpair_kernel(
    const TX* __restrict__ x0, const TX* __restrict__ x1,
    const __hip_bfloat16* __restrict__ w1,
    const __hip_bfloat16* __restrict__ b1,
    const __hip_bfloat16* __restrict__ w2,
    const __hip_bfloat16* __restrict__ b2,
    const __hip_bfloat16* __restrict__ w3,
    const __hip_bfloat16* __restrict__ b3, __hip_bfloat16* __restrict__ out0,
    __hip_bfloat16* __restrict__ out1, __hip_bfloat16* __restrict__ scratch,
    const int N, const int O, const int H, const int A2) {
  const int net = blockIdx.y;
  __hip_bfloat16* h1 = scratch + (size_t)net * 2 * N * H;
  __hip_bfloat16* h2 = h1 + (size_t)N * H;
  mlp3_mfma_fwd_impl<TX, R>(net ? x1 : x0, w1, b1, w2, b2, w3, b3,
                            net ? out1 : out0, h1, h2, nullptr, N, O, H, A2);
}

extern "C" void launch_mlp3_mfma_fwdpair(const void* x0, const void* x1,
                                         int x_is_f32, const void* const* w,
                                         void* out0, void* out1,
                                         void* scratch, long N, int O, int H,
                                         int A2, void* stream) {
  const int R = N >= M3_ROWS_SWITCH ? 64 : 32;
  const int blocks = (int)((N + R - 1) / R);
  const int Op = (O + 31) & ~31, Hp = (H + 31) & ~31;
  const int lx = Op + 8, lh = Hp + 8, A2p = (A2 + 15) & ~15;
  const int lds = 2 * (R * lx + H * lx + 2 * R * lh + H * lh + A2p * lh);
  dim3 grid(blocks, 2);
#define M3_LAUNCH_PAIR(TX, RR)                                              \
  hipLaunchKernelGGL((mlp3_mfma_fwdpair_kernel<TX, RR>), grid,              \
                     dim3(MLP_THREADS), lds, (hipStream_t)stream,           \
                     (const TX*)x0, (const TX*)x1,                          \
                     (const __hip_bfloat16*)w[0],                           \
                     (const __hip_bfloat16*)w[1],                           \
                     (const __hip_bfloat16*)w[2],                           \
                     (const __hip_bfloat16*)w[3],                           \
                     (const __hip_bfloat16*)w[4],                           \
                     (const __hip_bfloat16*)w[5], (__hip_bfloat16*)out0,    \
                     (__hip_bfloat16*)out1, (__hip_bfloat16*)scratch,       \
                     (int)N, O, H, A2)
  if (x_is_f32) {
    if (R == 64) M3_LAUNCH_PAIR(float, 64);
    else M3_LAUNCH_PAIR(float, 32);
  } else {
    if (R == 64) M3_LAUNCH_PAIR(__hip_bfloat16, 64);
    else M3_LAUNCH_PAIR(__hip_bfloat16, 32);
  }
#undef M3_LAUNCH_PAIR
}

// ---------------------------------------------------------------------------
// Dual-network kernels WITH the PPO loss fused in (the whole-minibatch
// endgame): grid.y = 0 runs the actor MLP and emits the clipped-PPO /
// entropy per-workgroup partial sums straight from its LDS-staged head
// rows; grid.y = 1 runs the critic and emits the smooth-L1 partial.
// One finalize writes the 8 loss scalars.  The backward twin computes
// d(head)/d(value) per row IN the dgrad kernel (writing them to global
// for the batched wgrad) before running the dgrad chain.  Loss math is
// copied 1:1 from csrc/loss_ops.hip (fast-math forms: both nets are
// bf16-quantized).
// ---------------------------------------------------------------------------

namespace {

__device__ __forceinline__ float ac_exp(float x) { return __expf(x); }
__device__ __forceinline__ float ac_log(float x) { return __logf(x); }
__device__ __forceinline__ float ac_tanh(float x) {
  const float cx = fminf(fmaxf(x, -15.f), 15.f);
  const float t = __expf(2.f * cx);
  return (t - 1.f) / (t + 1.f);
}
__device__ __forceinline__ float ac_atanh(float y) {
  return 0.5f * __logf((1.f + y) / (1.f - y));
}
__device__ __forceinline__ float ac_softplus(float x) {
  if (x > 20.f) return x;
  return __logf(1.f + __expf(x));
}

#define AC_LOG_SQRT_2PI 0.9189385332046727f
#define AC_LOG2 0.6931471805599453f
#define AC_ATANH_LIM (1.0f - 1.1920929e-7f)

struct ACLossArgs {
  const float* action;   // [N, A]
  const float* eps;      // [N, A]
  const float* prev;     // [N]
  const float* adv;      // [N]
  const float* stats;    // nullable (mu, 1/sigma)
  const float* vtarget;  // [N]
  float* part;           // [nWG, 7]: 0..4 actor sums (WG y=0), 5 crit,
                         // 6 unused
  float sp_bias, lb, lo, hi;
  int Aact;
};

__device__ __forceinline__ float ac_block_sum(float v, float* smem) {
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
  const int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  if (lane == 0) smem[wave] = v;
  __syncthreads();
  if (wave == 0) {
    v = (lane < (int)(blockDim.x >> 6)) ? smem[lane] : 0.f;
    for (int off = 2; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
  }
  return v;
}

template <typename TX, int R>
__global__ void __launch_bounds__(MLP_THREADS) mlp3_mfma_fwd2_loss_kernel(
    const TX* __restrict__ x, const AC2Fwd nets, const ACLossArgs loss,
    __hip_bfloat16* __restrict__ xb_out, const int N, const int O) {
  const int net = blockIdx.y;
  mlp3_mfma_fwd_impl<TX, R>(x, nets.w1[net], nets.b1[net], nets.w2[net],
                            nets.b2[net], nets.w3[net], nets.b3[net],
                            nets.out[net], nets.h1[net], nets.h2[net],
                            net == 0 ? xb_out : nullptr, N, O, nets.H[net],
                            nets.A2[net]);
  // the head/value rows this WG just produced are still LDS-staged in
  // the impl's s_h1 region (same extern-smem layout math as the impl)
  const int H = nets.H[net];
  const int Op = m3_pad32(O), Hp = m3_pad32(H);
  const int lx = Op + 8, lh = Hp + 8;
  extern __shared__ __hip_bfloat16 smem[];
  const __hip_bfloat16* s_head = smem + (size_t)R * lx + (size_t)H * lx;
  const long row0 = (long)blockIdx.x * R;
  const int rows = (int)min((long)R, (long)N - row0);
  const int tid = threadIdx.x;
  __shared__ float red[8];
  const int A = loss.Aact;
  if (net == 1) {
    // critic: smooth-L1 partial over this WG's rows
    float sv = 0.f;
    for (int r = tid; r < rows; r += MLP_THREADS) {
      const float z =
          __bfloat162float(s_head[(size_t)r * lh]) - loss.vtarget[row0 + r];
      const float az = fabsf(z);
      sv += (az < 1.f) ? 0.5f * z * z : az - 0.5f;
    }
    const float t = ac_block_sum(sv, red);
    if (tid == 0) loss.part[(size_t)blockIdx.x * 7 + 5] = t;
    return;
  }
  // actor: clipped surrogate + entropy partials from the staged head
  const float mu = loss.stats ? loss.stats[0] : 0.f;
  const float isd = loss.stats ? loss.stats[1] : 1.f;
  float sg = 0.f, sr = 0.f, sr2 = 0.f, sc = 0.f, se = 0.f;
  for (int r = tid; r < rows; r += MLP_THREADS) {
    const long n = row0 + r;
    float lp = 0.f, ent = 0.f;
    for (int a = 0; a < A; ++a) {
      const float loc = __bfloat162float(s_head[(size_t)r * lh + a]);
      const float spre =
          __bfloat162float(s_head[(size_t)r * lh + A + a]) + loss.sp_bias;
      const float sv_ = fmaxf(ac_softplus(spre), loss.lb);
      const float y = fminf(fmaxf(loss.action[n * A + a], -AC_ATANH_LIM),
                            AC_ATANH_LIM);
      const float u = ac_atanh(y);
      const float z = (u - loc) / sv_;
      lp += -0.5f * z * z - ac_log(sv_) - AC_LOG_SQRT_2PI
            - 2.0f * (AC_LOG2 - u - ac_softplus(-2.0f * u));
      const float e = loss.eps[n * A + a];
      const float xx = ac_tanh(loc + sv_ * e);
      ent += 0.5f * e * e + ac_log(sv_) + AC_LOG_SQRT_2PI + log1pf(-xx * xx);
    }
    const float w = lp - loss.prev[n];
    const float an = (loss.adv[n] - mu) * isd;
    const float rr = ac_exp(w);
    const float rc = ac_exp(fminf(fmaxf(w, loss.lo), loss.hi));
    sg += fminf(rr * an, rc * an);
    sr += rr;
    sr2 += ac_exp(2.0f * w);
    sc += (rc != rr) ? 1.f : 0.f;
    se += ent;
  }
  float t;
  t = ac_block_sum(sg, red);
  if (tid == 0) loss.part[(size_t)blockIdx.x * 7 + 0] = t;
  __syncthreads();
  t = ac_block_sum(sr, red);
  if (tid == 0) loss.part[(size_t)blockIdx.x * 7 + 1] = t;
  __syncthreads();
  t = ac_block_sum(sr2, red);
  if (tid == 0) loss.part[(size_t)blockIdx.x * 7 + 2] = t;
  __syncthreads();
  t = ac_block_sum(sc, red);
  if (tid == 0) loss.part[(size_t)blockIdx.x * 7 + 3] = t;
  __syncthreads();
  t = ac_block_sum(se, red);
  if (tid == 0) loss.part[(size_t)blockIdx.x * 7 + 4] = t;
}

struct ACOuts {
  float* o[8];
};

__global__ void acloss_finalize_k(const float* __restrict__ part,
                                  const int nwg, const long N,
                                  const float ent_coeff,
                                  const float crit_scale,
                                  const ACOuts outs_) {
  float* const* outs = outs_.o;
  float sg = 0.f, sr = 0.f, sr2 = 0.f, sc = 0.f, se = 0.f, sv = 0.f;
  for (int i = threadIdx.x; i < nwg; i += blockDim.x) {
    sg += part[7 * i];
    sr += part[7 * i + 1];
    sr2 += part[7 * i + 2];
    sc += part[7 * i + 3];
    se += part[7 * i + 4];
    sv += part[7 * i + 5];
  }
  __shared__ float smem[8];
  float t;
  t = ac_block_sum(sg, smem);
  if (threadIdx.x == 0) smem[4] = t;
  __syncthreads();
  const float tsg = smem[4];
  t = ac_block_sum(sr, smem);
  if (threadIdx.x == 0) smem[4] = t;
  __syncthreads();
  const float tsr = smem[4];
  t = ac_block_sum(sr2, smem);
  if (threadIdx.x == 0) smem[4] = t;
  __syncthreads();
  const float tsr2 = smem[4];
  t = ac_block_sum(sc, smem);
  if (threadIdx.x == 0) smem[4] = t;
  __syncthreads();
  const float tsc = smem[4];
  t = ac_block_sum(se, smem);
  if (threadIdx.x == 0) smem[4] = t;
  __syncthreads();
  const float tse = smem[4];
  t = ac_block_sum(sv, smem);
  if (threadIdx.x == 0) {
    const float loss_obj = -tsg / (float)N;
    *outs[0] = loss_obj;
    *outs[1] = tsr * tsr / fmaxf(tsr2, 1e-12f) / (float)N;  // ESS/N
    *outs[2] = tsc / (float)N;                              // clip_fraction
    *outs[3] = tse / (float)N;                              // entropy
    const float lent = -ent_coeff * tse / (float)N;
    *outs[4] = lent;                                        // loss_entropy
    const float lact = loss_obj + lent;
    *outs[5] = lact;                                        // loss_actor
    const float lcrit = crit_scale * t / (float)N;
    *outs[6] = lcrit;                                       // loss_critic
    *outs[7] = lact + lcrit;                                // total
  }
}

struct ACLossBwdArgs {
  const float* action;
  const float* eps;
  const float* prev;
  const float* adv;
  const float* stats;
  const float* vtarget;
  const float* gobj;   // nullable 0-d grads
  const float* gent;
  const float* gact;
  const float* gcrit;
  const float* gtot;
  float sp_bias, lb, lo, hi, ent_coeff, crit_scale;
  int Aact;
};

// dhead/dvalue computed per row into the GLOBAL dout buffers (the
// batched wgrad reads them), then the standard dgrad chain runs.
template <int R>
__global__ void __launch_bounds__(MLP_THREADS) mlp3_mfma_bwd2_loss_kernel(
    const AC2Bwd nets, const ACLossBwdArgs a,
    __hip_bfloat16* __restrict__ dhead_out,   // [N, A2_actor]
    __hip_bfloat16* __restrict__ dvalue_out,  // [N, 1]
    const __hip_bfloat16* __restrict__ head,  // [N, A2_actor]
    const __hip_bfloat16* __restrict__ value, // [N, 1]
    const int N) {
  const int net = blockIdx.y;
  const long row0 = (long)blockIdx.x * R;
  const int rows = (int)min((long)R, (long)N - row0);
  const int tid = threadIdx.x;
  const int A = a.Aact;
  const float gt = a.gtot ? a.gtot[0] : 0.f;
  if (net == 1) {
    const float gvs = a.crit_scale *
                      ((a.gcrit ? a.gcrit[0] : 0.f) + gt) / (float)N;
    for (int r = tid; r < rows; r += MLP_THREADS) {
      const long n = row0 + r;
      const float z = __bfloat162float(value[n]) - a.vtarget[n];
      dvalue_out[n] = __hip_bfloat16(gvs * fminf(fmaxf(z, -1.f), 1.f));
    }
  } else {
    const float mu = a.stats ? a.stats[0] : 0.f;
    const float isd = a.stats ? a.stats[1] : 1.f;
    const float ga = (a.gact ? a.gact[0] : 0.f) + gt;
    const float g1s = -(ga + (a.gobj ? a.gobj[0] : 0.f)) / (float)N;
    const float ges =
        -a.ent_coeff * (ga + (a.gent ? a.gent[0] : 0.f)) / (float)N;
    for (int r = tid; r < rows; r += MLP_THREADS) {
      const long n = row0 + r;
      float lp = 0.f;
      for (int q = 0; q < A; ++q) {
        const float loc = __bfloat162float(head[n * 2 * A + q]);
        const float sv_ = fmaxf(
            ac_softplus(__bfloat162float(head[n * 2 * A + A + q]) +
                        a.sp_bias),
            a.lb);
        const float y = fminf(fmaxf(a.action[n * A + q], -AC_ATANH_LIM),
                              AC_ATANH_LIM);
        const float u = ac_atanh(y);
        const float z = (u - loc) / sv_;
        lp += -0.5f * z * z - ac_log(sv_) - AC_LOG_SQRT_2PI
              - 2.0f * (AC_LOG2 - u - ac_softplus(-2.0f * u));
      }
      const float w = lp - a.prev[n];
      const float an = (a.adv[n] - mu) * isd;
      const float rr = ac_exp(w);
      const float rc = ac_exp(fminf(fmaxf(w, a.lo), a.hi));
      const float dlw = (rr * an <= rc * an) ? g1s * an * rr : 0.f;
      for (int q = 0; q < A; ++q) {
        const float loc = __bfloat162float(head[n * 2 * A + q]);
        const float spre =
            __bfloat162float(head[n * 2 * A + A + q]) + a.sp_bias;
        const float s0 = ac_softplus(spre);
        const float sv_ = fmaxf(s0, a.lb);
        const float y = fminf(fmaxf(a.action[n * A + q], -AC_ATANH_LIM),
                              AC_ATANH_LIM);
        const float u = ac_atanh(y);
        const float z = (u - loc) / sv_;
        const float e = a.eps[n * A + q];
        const float xx = ac_tanh(loc + sv_ * e);
        const float dloc = dlw * z / sv_ + ges * (-2.0f * xx);
        float ds = dlw * (z * z - 1.0f) / sv_ +
                   ges * (1.0f / sv_ - 2.0f * xx * e);
        ds = (s0 >= a.lb) ? ds : 0.f;
        const float sig = 1.0f / (1.0f + __expf(-spre));
        dhead_out[n * 2 * A + q] = __hip_bfloat16(dloc);
        dhead_out[n * 2 * A + A + q] = __hip_bfloat16(ds * sig);
      }
    }
  }
  __syncthreads();  // the WG's own global dout rows are what it stages
  mlp3_mfma_bwd_impl<__hip_bfloat16, R>(
      net == 0 ? dhead_out : dvalue_out, nets.h1[net], nets.h2[net],
      nets.w2[net], nets.w3[net], nets.dh1[net], nets.dh2[net], N,
      nets.H[net], nets.A2[net]);
}

}  // namespace

extern "C" void launch_mlp3_mfma_fwd2_loss(
    const void* x, int x_is_f32, const void* const* w, void* const* o,
    void* xb, const float* action, const float* eps, const float* prev,
    const float* adv, const float* stats, const float* vtarget, float* part,
    float* const* outs, float sp_bias, float lb, float lo, float hi,
    float ent_coeff, float crit_scale, long N, int O, const int* H,
    const int* A2, int Aact, void* stream) {
  AC2Fwd nets;
  for (int n = 0; n < 2; ++n) {
    nets.w1[n] = (const __hip_bfloat16*)w[n * 6 + 0];
    nets.b1[n] = (const __hip_bfloat16*)w[n * 6 + 1];
    nets.w2[n] = (const __hip_bfloat16*)w[n * 6 + 2];
    nets.b2[n] = (const __hip_bfloat16*)w[n * 6 + 3];
    nets.w3[n] = (const __hip_bfloat16*)w[n * 6 + 4];
    nets.b3[n] = (const __hip_bfloat16*)w[n * 6 + 5];
    nets.out[n] = (__hip_bfloat16*)o[n * 3 + 0];
    nets.h1[n] = (__hip_bfloat16*)o[n * 3 + 1];
    nets.h2[n] = (__hip_bfloat16*)o[n * 3 + 2];
    nets.H[n] = H[n];
    nets.A2[n] = A2[n];
  }
  ACLossArgs la;
  la.action = action;
  la.eps = eps;
  la.prev = prev;
  la.adv = adv;
  la.stats = stats;
  la.vtarget = vtarget;
  la.part = part;
  la.sp_bias = sp_bias;
  la.lb = lb;
  la.lo = lo;
  la.hi = hi;
  la.Aact = Aact;
  const int R = N >= M3_ROWS_SWITCH ? 64 : 32;
  const int blocks = (int)((N + R - 1) / R);
  int lds = 0;
  for (int n = 0; n < 2; ++n) {
    const int Op = (O + 31) & ~31, Hp = (H[n] + 31) & ~31;
    const int lx = Op + 8, lh = Hp + 8, A2p = (A2[n] + 15) & ~15;
    const int b = 2 * (R * lx + H[n] * lx + 2 * R * lh + H[n] * lh + A2p * lh);
    if (b > lds) lds = b;
  }
  dim3 grid(blocks, 2);
  if (x_is_f32) {
    if (R == 64)
      hipLaunchKernelGGL((mlp3_mfma_fwd2_loss_kernel<float, 64>), grid,
                         dim3(MLP_THREADS), lds, (hipStream_t)stream,
                         (const float*)x, nets, la, (__hip_bfloat16*)xb,
                         (int)N, O);
    else
      hipLaunchKernelGGL((mlp3_mfma_fwd2_loss_kernel<float, 32>), grid,
                         dim3(MLP_THREADS), lds, (hipStream_t)stream,
                         (const float*)x, nets, la, (__hip_bfloat16*)xb,
                         (int)N, O);
  } else {
    if (R == 64)
      hipLaunchKernelGGL((mlp3_mfma_fwd2_loss_kernel<__hip_bfloat16, 64>),
                         grid, dim3(MLP_THREADS), lds, (hipStream_t)stream,
                         (const __hip_bfloat16*)x, nets, la,
                         (__hip_bfloat16*)xb, (int)N, O);
    else
      hipLaunchKernelGGL((mlp3_mfma_fwd2_loss_kernel<__hip_bfloat16, 32>),
                         grid, dim3(MLP_THREADS), lds, (hipStream_t)stream,
                         (const __hip_bfloat16*)x, nets, la,
                         (__hip_bfloat16*)xb, (int)N, O);
  }
  ACOuts ao;
  for (int i = 0; i < 8; ++i) ao.o[i] = outs[i];
  hipLaunchKernelGGL(acloss_finalize_k, dim3(1), dim3(MLP_THREADS), 0,
                     (hipStream_t)stream, part, blocks, N, ent_coeff,
                     crit_scale, ao);
}

extern "C" void launch_mlp3_mfma_bwd2_loss(
    const void* const* hw,  // 8: a_h1,a_h2,a_w2,a_w3,c_h1,c_h2,c_w2,c_w3
    void* const* dh,        // 4: a_dh1,a_dh2,c_dh1,c_dh2
    const float* action, const float* eps, const float* prev,
    const float* adv, const float* stats, const float* vtarget,
    const float* gobj, const float* gent, const float* gact,
    const float* gcrit, const float* gtot, void* dhead, void* dvalue,
    const void* head, const void* value, float sp_bias, float lb, float lo,
    float hi, float ent_coeff, float crit_scale, long N, const int* H,
    const int* A2, int Aact, void* stream) {
  AC2Bwd nets;
  for (int n = 0; n < 2; ++n) {
    nets.dout[n] = nullptr;
    nets.h1[n] = (const __hip_bfloat16*)hw[n * 4 + 0];
    nets.h2[n] = (const __hip_bfloat16*)hw[n * 4 + 1];
    nets.w2[n] = (const __hip_bfloat16*)hw[n * 4 + 2];
    nets.w3[n] = (const __hip_bfloat16*)hw[n * 4 + 3];
    nets.dh1[n] = (__hip_bfloat16*)dh[n * 2 + 0];
    nets.dh2[n] = (__hip_bfloat16*)dh[n * 2 + 1];
    nets.H[n] = H[n];
    nets.A2[n] = A2[n];
  }
  ACLossBwdArgs a;
  a.action = action;
  a.eps = eps;
  a.prev = prev;
  a.adv = adv;
  a.stats = stats;
  a.vtarget = vtarget;
  a.gobj = gobj;
  a.gent = gent;
  a.gact = gact;
  a.gcrit = gcrit;
  a.gtot = gtot;
  a.sp_bias = sp_bias;
  a.lb = lb;
  a.lo = lo;
  a.hi = hi;
  a.ent_coeff = ent_coeff;
  a.crit_scale = crit_scale;
  a.Aact = Aact;
  const int R = N >= M3_ROWS_SWITCH ? 64 : 32;
  const int blocks = (int)((N + R - 1) / R);
  int lds = 0;
  for (int n = 0; n < 2; ++n) {
    const int ld = ((A2[n] + 31) & ~31) + 8, lh = ((H[n] + 31) & ~31) + 8;
    const int b = 2 * (R * ld + H[n] * ld + 2 * R * lh + H[n] * lh);
    if (b > lds) lds = b;
  }
  dim3 grid(blocks, 2);
  if (R == 64)
    hipLaunchKernelGGL((mlp3_mfma_bwd2_loss_kernel<64>), grid,
                       dim3(MLP_THREADS), lds, (hipStream_t)stream, nets, a,
                       (__hip_bfloat16*)dhead, (__hip_bfloat16*)dvalue,
                       (const __hip_bfloat16*)head,
                       (const __hip_bfloat16*)value, (int)N);
  else
    hipLaunchKernelGGL((mlp3_mfma_bwd2_loss_kernel<32>), grid,
                       dim3(MLP_THREADS), lds, (hipStream_t)stream, nets, a,
                       (__hip_bfloat16*)dhead, (__hip_bfloat16*)dvalue,
                       (const __hip_bfloat16*)head,
                       (const __hip_bfloat16*)value, (int)N);
}
