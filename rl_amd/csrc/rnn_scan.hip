// Fused recurrent scans with per-step reset for CDNA4 (gfx950).
//
// Capability of the reference's Triton kernels
// (pytorch/rl torchrl/modules/tensordict_module/_rnn_triton.py:191
// _gru_fwd_kernel, :644 _lstm_fwd_kernel): the whole T-step recurrence is
// ONE kernel launch; the x@W_ih GEMM for all timesteps is done outside by
// one large GEMM (hipBLASLt via torch.matmul), and `is_init` zeroes the
// state at trajectory starts.
//
// MI355X design (not a port — the reference tiles for 32-wide warps and
// NVIDIA's 255-VGPR ceiling, see its comment block :159-177):
//  * one workgroup per B_TILE=32 batch rows; grid = ceil(B/32) * 1.
//  * W_hh^T staged ONCE in LDS as bf16 [H][G*H] (G=3 GRU / 4 LSTM);
//    H=128 GRU: 96 KiB, fits the 160 KiB LDS with the fp32 h tile.
//  * each thread owns (row, hidden-slot) pairs and computes ALL G gate
//    dot-products for its pair, so the r·gh_n product (GRU) and the
//    i,f,g,o combination (LSTM) are thread-local — no cross-lane traffic
//    in the update.
//  * W reads are LDS-coalesced (consecutive threads → consecutive j).
//
// Forward-only: collector rollouts run under no_grad; sequence training
// uses the autograd scan (rnn.py gru_scan/lstm_scan) which this kernel is
// numerics-checked against in tests/test_rnn.py.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define THREADS 256
#define B_TILE 8

namespace {

__device__ __forceinline__ float sigmoidf_(float x) {
  return 1.0f / (1.0f + __expf(-x));
}

// gates_x: [B, T, 3H] (= x@W_ih + bias_ih + bias_hh, biases folded on host
// EXCEPT the n-gate's hidden bias which must stay inside r*(...)):
// here gates_x = x@W_ih + bias_ih only; bias_hh passed separately.
__global__ void gru_fused_kernel(
    const float* __restrict__ gates_x,  // [B, T, 3H]
    const float* __restrict__ w_hh,     // [3H, H] row-major
    const float* __restrict__ bias_hh,  // [3H]
    const bool* __restrict__ is_init,   // [B, T]
    const float* __restrict__ h0,       // [B, H] or nullptr
    float* __restrict__ ys,             // [B, T, H]
    float* __restrict__ h_out,          // [B, H]
    const int B, const int T, const int H) {
  extern __shared__ unsigned char smem[];
  // layout: W^T bf16 [H][3H], then h fp32 [B_TILE][H]
  __hip_bfloat16* w_t = reinterpret_cast<__hip_bfloat16*>(smem);
  float* h_s = reinterpret_cast<float*>(smem + (size_t)H * 3 * H * sizeof(__hip_bfloat16));

  const int tid = threadIdx.x;
  const int row0 = blockIdx.x * B_TILE;
  const int rows = min(B_TILE, B - row0);

  // stage W^T: w_t[k][j] = w_hh[j][k]
  for (int i = tid; i < 3 * H * H; i += THREADS) {
    const int j = i / H;   // output index in [0, 3H)
    const int k = i % H;   // input index in [0, H)
    w_t[(size_t)k * 3 * H + j] = __float2bfloat16(w_hh[(size_t)j * H + k]);
  }
  // stage h0
  for (int i = tid; i < rows * H; i += THREADS) {
    const int b = i / H;
    const int k = i % H;
    h_s[b * H + k] = h0 ? h0[(size_t)(row0 + b) * H + k] : 0.0f;
  }
  __syncthreads();

  const int pairs = rows * H;                 // (row, hidden-slot) pairs
  for (int t = 0; t < T; ++t) {
    // zero state at trajectory starts
    for (int i = tid; i < pairs; i += THREADS) {
      const int b = i / H;
      if (is_init[(size_t)(row0 + b) * T + t]) h_s[i] = 0.0f;
    }
    __syncthreads();
    // each thread: all 3 gate dots for its (b, jh) pairs
    for (int i = tid; i < pairs; i += THREADS) {
      const int b = i / H;
      const int jh = i % H;
      float acc_r = bias_hh[jh];
      float acc_z = bias_hh[H + jh];
      float acc_n = bias_hh[2 * H + jh];
      const float* hrow = &h_s[b * H];
      const __hip_bfloat16* wt = w_t;
      for (int k = 0; k < H; ++k) {
        const float hv = hrow[k];
        const size_t base = (size_t)k * 3 * H;
        acc_r += hv * __bfloat162float(wt[base + jh]);
        acc_z += hv * __bfloat162float(wt[base + H + jh]);
        acc_n += hv * __bfloat162float(wt[base + 2 * H + jh]);
      }
      const size_t gbase = ((size_t)(row0 + b) * T + t) * 3 * H;
      const float r = sigmoidf_(gates_x[gbase + jh] + acc_r);
      const float z = sigmoidf_(gates_x[gbase + H + jh] + acc_z);
      const float n = tanhf(gates_x[gbase + 2 * H + jh] + r * acc_n);
      const float hnew = (1.0f - z) * n + z * hrow[jh];
      ys[((size_t)(row0 + b) * T + t) * H + jh] = hnew;
      // defer the LDS write until all reads of h_s for this step are done
      // → stash in ys (already written) and sync below
    }
    __syncthreads();
    for (int i = tid; i < pairs; i += THREADS) {
      const int b = i / H;
      const int jh = i % H;
      h_s[i] = ys[((size_t)(row0 + b) * T + t) * H + jh];
    }
    __syncthreads();
  }
  for (int i = tid; i < pairs; i += THREADS) {
    const int b = i / H;
    const int k = i % H;
    h_out[(size_t)(row0 + b) * H + k] = h_s[i];
  }
}

// LSTM: gates order i, f, g, o (torch convention).
__global__ void lstm_fused_kernel(
    const float* __restrict__ gates_x,  // [B, T, 4H] = x@W_ih + b_ih + b_hh
    const float* __restrict__ w_hh,     // [4H, H]
    const bool* __restrict__ is_init,   // [B, T]
    const float* __restrict__ h0,       // [B, H] or nullptr
    const float* __restrict__ c0,       // [B, H] or nullptr
    float* __restrict__ ys,             // [B, T, H]
    float* __restrict__ h_out,          // [B, H]
    float* __restrict__ c_out,          // [B, H]
    const int B, const int T, const int H) {
  extern __shared__ unsigned char smem[];
  __hip_bfloat16* w_t = reinterpret_cast<__hip_bfloat16*>(smem);
  float* h_s = reinterpret_cast<float*>(smem + (size_t)H * 4 * H * sizeof(__hip_bfloat16));
  float* c_s = h_s + B_TILE * H;

  const int tid = threadIdx.x;
  const int row0 = blockIdx.x * B_TILE;
  const int rows = min(B_TILE, B - row0);

  for (int i = tid; i < 4 * H * H; i += THREADS) {
    const int j = i / H;
    const int k = i % H;
    w_t[(size_t)k * 4 * H + j] = __float2bfloat16(w_hh[(size_t)j * H + k]);
  }
  for (int i = tid; i < rows * H; i += THREADS) {
    const int b = i / H;
    const int k = i % H;
    h_s[b * H + k] = h0 ? h0[(size_t)(row0 + b) * H + k] : 0.0f;
    c_s[b * H + k] = c0 ? c0[(size_t)(row0 + b) * H + k] : 0.0f;
  }
  __syncthreads();

  const int pairs = rows * H;
  for (int t = 0; t < T; ++t) {
    for (int i = tid; i < pairs; i += THREADS) {
      const int b = i / H;
      if (is_init[(size_t)(row0 + b) * T + t]) {
        h_s[i] = 0.0f;
        c_s[i] = 0.0f;
      }
    }
    __syncthreads();
    for (int i = tid; i < pairs; i += THREADS) {
      const int b = i / H;
      const int jh = i % H;
      float acc_i = 0.f, acc_f = 0.f, acc_g = 0.f, acc_o = 0.f;
      const float* hrow = &h_s[b * H];
      for (int k = 0; k < H; ++k) {
        const float hv = hrow[k];
        const size_t base = (size_t)k * 4 * H;
        acc_i += hv * __bfloat162float(w_t[base + jh]);
        acc_f += hv * __bfloat162float(w_t[base + H + jh]);
        acc_g += hv * __bfloat162float(w_t[base + 2 * H + jh]);
        acc_o += hv * __bfloat162float(w_t[base + 3 * H + jh]);
      }
      const size_t gbase = ((size_t)(row0 + b) * T + t) * 4 * H;
      const float ig = sigmoidf_(gates_x[gbase + jh] + acc_i);
      const float fg = sigmoidf_(gates_x[gbase + H + jh] + acc_f);
      const float gg = tanhf(gates_x[gbase + 2 * H + jh] + acc_g);
      const float og = sigmoidf_(gates_x[gbase + 3 * H + jh] + acc_o);
      const float cnew = fg * c_s[b * H + jh] + ig * gg;
      const float hnew = og * tanhf(cnew);
      ys[((size_t)(row0 + b) * T + t) * H + jh] = hnew;
      // cnew stashed in c via two-phase write below — store temporarily
      // in registers is impossible across the strided loop, so write c to
      // global scratch h_out (reused per step) then copy back
      c_out[(size_t)(row0 + b) * H + jh] = cnew;
    }
    __syncthreads();
    for (int i = tid; i < pairs; i += THREADS) {
      const int b = i / H;
      const int jh = i % H;
      h_s[i] = ys[((size_t)(row0 + b) * T + t) * H + jh];
      c_s[i] = c_out[(size_t)(row0 + b) * H + jh];
    }
    __syncthreads();
  }
  for (int i = tid; i < pairs; i += THREADS) {
    const int b = i / H;
    const int k = i % H;
    h_out[(size_t)(row0 + b) * H + k] = h_s[i];
  }
}

}  // namespace

extern "C" {

int gru_fused_lds_bytes(int H) {
  return (int)((size_t)H * 3 * H * sizeof(__hip_bfloat16) +
               (size_t)B_TILE * H * sizeof(float));
}

int lstm_fused_lds_bytes(int H) {
  return (int)((size_t)H * 4 * H * sizeof(__hip_bfloat16) +
               (size_t)2 * B_TILE * H * sizeof(float));
}

void launch_gru_fused(const float* gates_x, const float* w_hh,
                      const float* bias_hh, const bool* is_init,
                      const float* h0, float* ys, float* h_out, int B, int T,
                      int H, void* stream) {
  const int blocks = (B + B_TILE - 1) / B_TILE;
  const int lds = gru_fused_lds_bytes(H);
  hipLaunchKernelGGL(gru_fused_kernel, dim3(blocks), dim3(THREADS), lds,
                     (hipStream_t)stream, gates_x, w_hh, bias_hh, is_init, h0,
                     ys, h_out, B, T, H);
}

void launch_lstm_fused(const float* gates_x, const float* w_hh,
                       const bool* is_init, const float* h0, const float* c0,
                       float* ys, float* h_out, float* c_out, int B, int T,
                       int H, void* stream) {
  const int blocks = (B + B_TILE - 1) / B_TILE;
  const int lds = lstm_fused_lds_bytes(H);
  hipLaunchKernelGGL(lstm_fused_kernel, dim3(blocks), dim3(THREADS), lds,
                     (hipStream_t)stream, gates_x, w_hh, is_init, h0, c0, ys,
                     h_out, c_out, B, T, H);
}

}  // extern "C"
