// Split-K weight-gradient kernel for skinny linear layers (CDNA4, gfx950).
//
// Motivation (measured, rocprofv3 r11 profile of the PPO bench): for the
// backward of an MLP layer with B=16384 rows and 64x64 weights, hipBLASLt
// picks a single-workgroup MT64x64x256 kernel with NO split-K (GSU0) —
// one CU of 256 does the whole K=16384 reduction at ~101 us, 12.9% of the
// whole training step.  The op is memory/launch bound (B*(M+N)*2 bytes
// read; the FLOPs are trivial), so the right MI355X mapping is:
//
//   dW[n][m] = sum_k dY[k][n] * X[k][m]      (N = out, M = in features)
//
//  * grid.x = K/K_SLAB workgroups (hundreds — fills the 8 XCDs),
//    grid.y/z = N/64 x M/64 output tiles (1 for MLP-sized layers).
//  * each workgroup stages 64-row chunks of X and dY in LDS
//    (coalesced ushort2 loads), then each of the 256 threads owns a
//    4x4 output tile: 16 VALU FMAs per k-row from LDS broadcasts.
//  * fp32 accumulation in registers; each workgroup writes its partial
//    tile to a [slabs, N, M] buffer (NO atomics — measured: atomicAdd
//    with 256 contending workgroups per address serialized at ~49 us;
//    partials + a reduce kernel run in a fraction of that), then a
//    second kernel reduces over the slab axis.
//  * dBias[n] = sum_k dY[k][n] is fused (threads with tm==0).
//
// Two compute paths share the same LDS staging and partials layout:
//  * wgrad_mfma_kernel — MFMA path (mfma_f32_16x16x32_bf16): each of
//    the 4 waves owns 4 of the 16 16x16 output tiles; fragments are
//    read straight from the padded LDS chunk.  Fragment mapping
//    verified on hardware (benchmarks/mfma_probe.hip): A row/B col =
//    lane&15, C/D row = (lane>>4)*4+reg, col = lane&15; the k
//    enumeration only needs to be CONSISTENT between A and B (any
//    k-permutation cancels in the contraction).
//  * wgrad_splitk_kernel — VALU fallback for shapes that are not
//    multiples of 16 in spirit (still correct for any N, M via the
//    zero-padded staging; kept for comparison runs).
//
// Reference behavior: torch.nn.functional.linear backward (wgrad);
// numerics validated vs fp32 torch.mm in tests/test_ops.py.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

using bf16_frag = __attribute__((ext_vector_type(8))) short;
using f32_frag = __attribute__((ext_vector_type(4))) float;
using f32_frag16 = __attribute__((ext_vector_type(16))) float;

#define WG_THREADS 256
#define ROW_CHUNK 64   // k-rows staged in LDS per iteration
// K_SLAB is chosen at launch: ~128 slabs regardless of K, so the reduce
// stays cheap and the partials buffer bounded.

namespace {

__global__ void __launch_bounds__(WG_THREADS) wgrad_splitk_kernel(
    const __hip_bfloat16* __restrict__ dy,  // [K, N] row-major
    const __hip_bfloat16* __restrict__ x,   // [K, M] row-major
    float* __restrict__ dw,                 // partials, [tile, elem, slab]
    float* __restrict__ dbias,              // bias partials [n, slab]
    long K, int N, int M, int k_slab) {
  // output tile assigned to this workgroup
  const int n0 = blockIdx.y * 64;
  const int m0 = blockIdx.z * 64;
  const long k_begin = (long)blockIdx.x * k_slab;
  const long k_end = min(K, k_begin + (long)k_slab);

  // thread's 4x4 sub-tile
  const int tn = threadIdx.x / 16;  // 0..15
  const int tm = threadIdx.x % 16;  // 0..15
  const int n_base = n0 + tn * 4;
  const int m_base = m0 + tm * 4;

  __shared__ float s_dy[ROW_CHUNK][64 + 1];  // +1 pad: stride 65 avoids bank conflicts
  __shared__ float s_x[ROW_CHUNK][64 + 1];

  float acc[4][4] = {};
  float bias_acc[4] = {};

  for (long kc = k_begin; kc < k_end; kc += ROW_CHUNK) {
    const int rows = (int)min((long)ROW_CHUNK, k_end - kc);
    // cooperative stage: 256 threads load rows*64 elements of each matrix.
    // thread t loads element (t/64 + 4*i, t%64) — fully coalesced.
    const int lr = threadIdx.x / 64;  // 0..3
    const int lc = threadIdx.x % 64;
    for (int i = 0; i < ROW_CHUNK / 4; ++i) {
      const int r = lr + 4 * i;
      if (r < rows) {
        const long gk = kc + r;
        s_dy[r][lc] = (lc < N - n0 && lc < 64)
                          ? __bfloat162float(dy[gk * N + n0 + lc])
                          : 0.0f;
        s_x[r][lc] = (lc < M - m0 && lc < 64)
                         ? __bfloat162float(x[gk * M + m0 + lc])
                         : 0.0f;
      } else {
        s_dy[r][lc] = 0.0f;
        s_x[r][lc] = 0.0f;
      }
    }
    __syncthreads();
    for (int k = 0; k < rows; ++k) {
      float dyv[4], xv[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) dyv[i] = s_dy[k][tn * 4 + i];
#pragma unroll
      for (int j = 0; j < 4; ++j) xv[j] = s_x[k][tm * 4 + j];
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j) acc[i][j] = fmaf(dyv[i], xv[j], acc[i][j]);
      if (tm == 0) {
#pragma unroll
        for (int i = 0; i < 4; ++i) bias_acc[i] += dyv[i];
      }
    }
    __syncthreads();
  }

  // write this workgroup's partial tile, slab-axis OUTERMOST: the
  // reduce's inner slab loop then reads CONSECUTIVE addresses across
  // lanes (coalesced) each iteration.  (The slab-innermost variant gave
  // each lane its own 512 B run — uncoalesced, measured 26.6 us.)
  const long slabs = gridDim.x;
  const long num_tiles_m = gridDim.z;
  const long tile_elems = (long)64 * 64 * gridDim.y * num_tiles_m;
  float* part = dw + (long)blockIdx.x * tile_elems +
                ((long)blockIdx.y * num_tiles_m + blockIdx.z) * (64 * 64);
#pragma unroll
  for (int i = 0; i < 4; ++i) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      part[(long)(tn * 4 + i) * 64 + tm * 4 + j] = acc[i][j];
    }
    if (dbias != nullptr && tm == 0 && m0 == 0) {
      dbias[(long)blockIdx.x * (gridDim.y * 64) + n_base + i] = bias_acc[i];
    }
  }
}

// MFMA split-K wgrad: same grid/staging/partials contract as the VALU
// kernel; compute runs on the matrix cores.
__global__ void __launch_bounds__(WG_THREADS) wgrad_mfma_kernel(
    const __hip_bfloat16* __restrict__ dy,  // [K, N]
    const __hip_bfloat16* __restrict__ x,   // [K, M]
    float* __restrict__ dw,                 // partials [slab, tile, elem]
    float* __restrict__ dbias,              // bias partials [slab, n]
    long K, int N, int M, int k_slab) {
  const int n0t = blockIdx.y * 64;
  const int m0t = blockIdx.z * 64;
  const long k_begin = (long)blockIdx.x * k_slab;
  const long k_end = min(K, k_begin + (long)k_slab);

  // bf16 LDS staging, DOUBLE-BUFFERED (pad: stride 66 shorts = 33 banks):
  // the next chunk's global loads land in the idle buffer while MFMA
  // consumes the current one, hiding the HBM latency.
  __shared__ __hip_bfloat16 s_dy[2][ROW_CHUNK][64 + 2];
  __shared__ __hip_bfloat16 s_x[2][ROW_CHUNK][64 + 2];
  __shared__ float s_bias[64];

  const int tid = threadIdx.x;
  const int wave = tid >> 6;       // 0..3
  const int lane = tid & 63;
  // wave w owns output tiles (n_tile, m_tile) = (w, 0..3): n rows
  // n0t + w*16 .. +15, all four 16-col m tiles.
  f32_frag acc[4] = {};
  float bias_acc = 0.f;

  const int lr = tid / 64;  // staging: thread t loads column t%64
  const int lc = tid % 64;

  auto stage = [&](long kc, int buf) {
    const int rows = (int)min((long)ROW_CHUNK, k_end - kc);
    for (int i = 0; i < ROW_CHUNK / 4; ++i) {
      const int r = lr + 4 * i;
      const long gk = kc + r;
      __hip_bfloat16 dv = __hip_bfloat16(0.f), xv = __hip_bfloat16(0.f);
      if (r < rows) {
        if (lc < N - n0t) dv = dy[gk * N + n0t + lc];
        if (lc < M - m0t) xv = x[gk * M + m0t + lc];
      }
      s_dy[buf][r][lc] = dv;
      s_x[buf][r][lc] = xv;
      bias_acc += __bfloat162float(dv);
    }
  };

  int cur = 0;
  stage(k_begin, 0);
  __syncthreads();
  for (long kc = k_begin; kc < k_end; kc += ROW_CHUNK) {
    if (kc + ROW_CHUNK < k_end) stage(kc + ROW_CHUNK, 1 - cur);
    // MFMA over the chunk: A[mf][k] = dy[k][n0w + mf], B[k][nf] = x[k][m0 + nf]
    const int n_off = wave * 16 + (lane & 15);
    for (int kk = 0; kk < ROW_CHUNK; kk += 32) {
      bf16_frag a;
#pragma unroll
      for (int reg = 0; reg < 8; ++reg) {
        const int k = kk + 8 * (lane >> 4) + reg;
        a[reg] = *reinterpret_cast<const short*>(&s_dy[cur][k][n_off]);
      }
#pragma unroll
      for (int mt = 0; mt < 4; ++mt) {
        bf16_frag b;
        const int m_off = mt * 16 + (lane & 15);
#pragma unroll
        for (int reg = 0; reg < 8; ++reg) {
          const int k = kk + 8 * (lane >> 4) + reg;
          b[reg] = *reinterpret_cast<const short*>(&s_x[cur][k][m_off]);
        }
        acc[mt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[mt], 0, 0, 0);
      }
    }
    __syncthreads();
    cur = 1 - cur;
  }

  // commit: D row = (lane>>4)*4 + reg (within the wave's 16-row band)
  const long num_tiles_m = gridDim.z;
  const long tile_elems = (long)64 * 64 * gridDim.y * num_tiles_m;
  float* part = dw + (long)blockIdx.x * tile_elems +
                ((long)blockIdx.y * num_tiles_m + blockIdx.z) * (64 * 64);
#pragma unroll
  for (int mt = 0; mt < 4; ++mt) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int n_loc = wave * 16 + (lane >> 4) * 4 + reg;
      const int m_loc = mt * 16 + (lane & 15);
      part[(long)n_loc * 64 + m_loc] = acc[mt][reg];
    }
  }
  // bias: 4 threads share each column; reduce through LDS
  if (dbias != nullptr) {
    if (lr == 0) s_bias[lc] = 0.f;
    __syncthreads();
    atomicAdd(&s_bias[lc], bias_acc);
    __syncthreads();
    if (lr == 0)
      dbias[(long)blockIdx.x * (gridDim.y * 64) + blockIdx.y * 64 + lc] = s_bias[lc];
  }
}

// 32x32-tile MFMA variant: each of the 4 waves owns ONE 32x32 output
// tile (mfma_f32_32x32x16_bf16; layout hardware-verified in
// benchmarks/mfma_probe.hip: A row / B col = lane&31,
// k = 8*(lane>>5)+reg, C/D col = lane&31,
// row = (reg&3)+8*(reg>>2)+4*(lane>>5)).  2.5x fewer LDS fragment
// reads per MFMA than the 16x16 tiling.
__global__ void __launch_bounds__(WG_THREADS) wgrad_mfma32_kernel(
    const __hip_bfloat16* __restrict__ dy,  // [K, N]
    const __hip_bfloat16* __restrict__ x,   // [K, M]
    float* __restrict__ dw,                 // partials [slab, tile, elem]
    float* __restrict__ dbias,              // bias partials [slab, n]
    long K, int N, int M, int k_slab) {
  const int n0t = blockIdx.y * 64;
  const int m0t = blockIdx.z * 64;
  const long k_begin = (long)blockIdx.x * k_slab;
  const long k_end = min(K, k_begin + (long)k_slab);

  __shared__ __hip_bfloat16 s_dy[2][ROW_CHUNK][64 + 2];
  __shared__ __hip_bfloat16 s_x[2][ROW_CHUNK][64 + 2];
  __shared__ float s_bias[64];

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int tile_n = (wave >> 1) * 32;
  const int tile_m = (wave & 1) * 32;
  f32_frag16 acc = {};
  float bias_acc = 0.f;

  const int lr = tid / 64;
  const int lc = tid % 64;

  auto stage = [&](long kc, int buf) {
    const int rows = (int)min((long)ROW_CHUNK, k_end - kc);
    for (int i = 0; i < ROW_CHUNK / 4; ++i) {
      const int r = lr + 4 * i;
      const long gk = kc + r;
      __hip_bfloat16 dv = __hip_bfloat16(0.f), xv = __hip_bfloat16(0.f);
      if (r < rows) {
        if (lc < N - n0t) dv = dy[gk * N + n0t + lc];
        if (lc < M - m0t) xv = x[gk * M + m0t + lc];
      }
      s_dy[buf][r][lc] = dv;
      s_x[buf][r][lc] = xv;
      bias_acc += __bfloat162float(dv);
    }
  };

  int cur = 0;
  stage(k_begin, 0);
  __syncthreads();
  const int n_off = tile_n + (lane & 31);
  const int m_off = tile_m + (lane & 31);
  for (long kc = k_begin; kc < k_end; kc += ROW_CHUNK) {
    if (kc + ROW_CHUNK < k_end) stage(kc + ROW_CHUNK, 1 - cur);
    for (int kk = 0; kk < ROW_CHUNK; kk += 16) {
      bf16_frag a, b;
#pragma unroll
      for (int reg = 0; reg < 8; ++reg) {
        const int k = kk + 8 * (lane >> 5) + reg;
        a[reg] = *reinterpret_cast<const short*>(&s_dy[cur][k][n_off]);
        b[reg] = *reinterpret_cast<const short*>(&s_x[cur][k][m_off]);
      }
      acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc, 0, 0, 0);
    }
    __syncthreads();
    cur = 1 - cur;
  }

  const long num_tiles_m = gridDim.z;
  const long tile_elems = (long)64 * 64 * gridDim.y * num_tiles_m;
  float* part = dw + (long)blockIdx.x * tile_elems +
                ((long)blockIdx.y * num_tiles_m + blockIdx.z) * (64 * 64);
#pragma unroll
  for (int reg = 0; reg < 16; ++reg) {
    const int n_loc = tile_n + (reg & 3) + 8 * (reg >> 2) + 4 * (lane >> 5);
    const int m_loc = tile_m + (lane & 31);
    part[(long)n_loc * 64 + m_loc] = acc[reg];
  }
  if (dbias != nullptr) {
    if (lr == 0) s_bias[lc] = 0.f;
    __syncthreads();
    atomicAdd(&s_bias[lc], bias_acc);
    __syncthreads();
    if (lr == 0)
      dbias[(long)blockIdx.x * (gridDim.y * 64) + blockIdx.y * 64 + lc] = s_bias[lc];
  }
}

// reduce the [slab, tile, elem] partials into dW [N, M]: one WAVE per
// output element — each lane sums slabs/64 strided values (all loads
// independent and in flight), then a 6-step shuffle tree; lane 0 writes
// without atomics.  (Thread-per-element was latency-bound at 26-56 us;
// thread-group + atomicAdd left only 2 waves/CU in flight.)
__global__ void wgrad_reduce_kernel(const float* __restrict__ part,
                                    const float* __restrict__ bias_part,
                                    float* __restrict__ dw,
                                    float* __restrict__ dbias, int slabs,
                                    int tiles_n, int tiles_m, int N, int M) {
  const long tile_elems = (long)64 * 64 * tiles_n * tiles_m;
  const long NM = (long)N * M;
  const long waves = ((long)gridDim.x * blockDim.x) >> 6;
  const long wave_id = ((long)blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  const int lane = threadIdx.x & 63;
  for (long e = wave_id; e < NM; e += waves) {
    const int n = (int)(e / M), m = (int)(e % M);
    const int tn = n / 64, tm = m / 64;
    const long off = ((long)tn * tiles_m + tm) * (64 * 64) +
                     (long)(n % 64) * 64 + (m % 64);
    float acc = 0.f;
    for (int s = lane; s < slabs; s += 64)
      acc += part[(long)s * tile_elems + off];
#pragma unroll
    for (int d = 32; d > 0; d >>= 1) acc += __shfl_down(acc, d);
    if (lane == 0) dw[e] = acc;
  }
  if (dbias != nullptr) {
    const long bias_waves = waves;
    for (long n = wave_id; n < N; n += bias_waves) {
      float acc = 0.f;
      for (int s = lane; s < slabs; s += 64)
        acc += bias_part[(long)s * (tiles_n * 64) + n];
#pragma unroll
      for (int d = 32; d > 0; d >>= 1) acc += __shfl_down(acc, d);
      if (lane == 0) dbias[n] = acc;
    }
  }
}

}  // namespace

extern "C" int wgrad_slab_count(long K) {
  // target ~256 slabs (one per CU): slab size a multiple of ROW_CHUNK.
  // (128 slabs left half the chip idle at K=65536 — measured 46 us.)
  long k_slab = (K + 255) / 256;
  k_slab = ((k_slab + ROW_CHUNK - 1) / ROW_CHUNK) * ROW_CHUNK;
  if (k_slab < ROW_CHUNK) k_slab = ROW_CHUNK;
  return (int)((K + k_slab - 1) / k_slab);
}

extern "C" void launch_wgrad_splitk(const void* dy, const void* x, float* dw,
                         float* dbias, float* part, float* bias_part, long K,
                         int N, int M, void* stream) {
  long k_slab = (K + 255) / 256;
  k_slab = ((k_slab + ROW_CHUNK - 1) / ROW_CHUNK) * ROW_CHUNK;
  if (k_slab < ROW_CHUNK) k_slab = ROW_CHUNK;
  const int slabs = (int)((K + k_slab - 1) / k_slab);
  const int tiles_n = (N + 63) / 64, tiles_m = (M + 63) / 64;
  dim3 grid(slabs, tiles_n, tiles_m);
  hipLaunchKernelGGL(wgrad_mfma32_kernel, grid, dim3(WG_THREADS), 0,
                     (hipStream_t)stream,
                     (const __hip_bfloat16*)dy, (const __hip_bfloat16*)x, part,
                     bias_part, K, N, M, (int)k_slab);
  const long total = (long)N * M * 64;  // one wave per element
  const int blocks = (int)min((total + 255) / 256, (long)2048);
  hipLaunchKernelGGL(wgrad_reduce_kernel, dim3(blocks), dim3(256), 0,
                     (hipStream_t)stream, part, bias_part, dw, dbias, slabs,
                     tiles_n, tiles_m, N, M);
}

// ---------------------------------------------------------------------------
// Batched 3-layer wgrad: the three (dY, X) weight-gradient pairs of one
// fused-MLP backward (layers emitted together by mlp3_mfma_bwd) in ONE
// mfma launch + ONE reduce launch instead of six.  All three layers
// share K (the minibatch rows); dims are <= 64 (one output tile each),
// which is exactly the PPO/critic MLP shape.  blockIdx.y selects the
// layer; the slab target halves vs the single-layer path (grid is 3x
// wider, so ~128 slabs x 3 layers still covers the 256 CUs) — the
// partials buffer and its reduce traffic halve with it.
// ---------------------------------------------------------------------------

struct Wg3Args {
  const __hip_bfloat16* dy[6];
  const __hip_bfloat16* x[6];
  float* part[6];       // [slabs, 64*64] each
  float* bias_part[6];  // [slabs, 64] each
  float* dw[6];
  float* db[6];
  int N[6];
  int M[6];
};

namespace {

__global__ void __launch_bounds__(WG_THREADS) wgrad3_mfma_kernel(
    const Wg3Args args, long K, int k_slab) {
  const int l = blockIdx.y;
  const __hip_bfloat16* __restrict__ dy = args.dy[l];
  const __hip_bfloat16* __restrict__ x = args.x[l];
  const int N = args.N[l], M = args.M[l];
  const long k_begin = (long)blockIdx.x * k_slab;
  const long k_end = min(K, k_begin + (long)k_slab);

  // row stride 72 shorts = 144 B: 16-byte aligned rows so full chunks
  // stage with uint4 (8 bf16) loads/stores when the layer is 64 wide
  __shared__ __hip_bfloat16 s_dy[2][ROW_CHUNK][64 + 8];
  __shared__ __hip_bfloat16 s_x[2][ROW_CHUNK][64 + 8];
  __shared__ float s_bias[64];

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int tile_n = (wave >> 1) * 32;
  const int tile_m = (wave & 1) * 32;
  f32_frag16 acc = {};
  float bias_acc = 0.f;     // scalar path: this thread's column lc
  float bias_v[8] = {};     // vector path: columns c8*8 .. c8*8+7
  const int c8 = tid & 7;   // fixed per thread (stride 256 keeps i&7)

  const int lr = tid / 64;
  const int lc = tid % 64;

  auto stage = [&](long kc, int buf) {
    const int rows = (int)min((long)ROW_CHUNK, k_end - kc);
    if (N == 64 && M == 64 && rows == ROW_CHUNK) {
      for (int i = tid; i < ROW_CHUNK * 8; i += WG_THREADS) {
        const int r = i >> 3;
        const uint4 dv =
            *reinterpret_cast<const uint4*>(&dy[(kc + r) * 64 + c8 * 8]);
        const uint4 xv =
            *reinterpret_cast<const uint4*>(&x[(kc + r) * 64 + c8 * 8]);
        *reinterpret_cast<uint4*>(&s_dy[buf][r][c8 * 8]) = dv;
        *reinterpret_cast<uint4*>(&s_x[buf][r][c8 * 8]) = xv;
        const __hip_bfloat16* hv =
            reinterpret_cast<const __hip_bfloat16*>(&dv);
#pragma unroll
        for (int j = 0; j < 8; ++j) bias_v[j] += __bfloat162float(hv[j]);
      }
      return;
    }
    for (int i = 0; i < ROW_CHUNK / 4; ++i) {
      const int r = lr + 4 * i;
      const long gk = kc + r;
      __hip_bfloat16 dv = __hip_bfloat16(0.f), xv = __hip_bfloat16(0.f);
      if (r < rows) {
        if (lc < N) dv = dy[gk * N + lc];
        if (lc < M) xv = x[gk * M + lc];
      }
      s_dy[buf][r][lc] = dv;
      s_x[buf][r][lc] = xv;
      bias_acc += __bfloat162float(dv);
    }
  };

  int cur = 0;
  stage(k_begin, 0);
  __syncthreads();
  const int n_off = tile_n + (lane & 31);
  const int m_off = tile_m + (lane & 31);
  for (long kc = k_begin; kc < k_end; kc += ROW_CHUNK) {
    if (kc + ROW_CHUNK < k_end) stage(kc + ROW_CHUNK, 1 - cur);
    for (int kk = 0; kk < ROW_CHUNK; kk += 16) {
      bf16_frag a, b;
#pragma unroll
      for (int reg = 0; reg < 8; ++reg) {
        const int k = kk + 8 * (lane >> 5) + reg;
        a[reg] = *reinterpret_cast<const short*>(&s_dy[cur][k][n_off]);
        b[reg] = *reinterpret_cast<const short*>(&s_x[cur][k][m_off]);
      }
      acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc, 0, 0, 0);
    }
    __syncthreads();
    cur = 1 - cur;
  }

  float* part = args.part[l] + (long)blockIdx.x * (64 * 64);
#pragma unroll
  for (int reg = 0; reg < 16; ++reg) {
    const int n_loc = tile_n + (reg & 3) + 8 * (reg >> 2) + 4 * (lane >> 5);
    const int m_loc = tile_m + (lane & 31);
    part[(long)n_loc * 64 + m_loc] = acc[reg];
  }
  if (lr == 0) s_bias[lc] = 0.f;
  __syncthreads();
  atomicAdd(&s_bias[lc], bias_acc);
#pragma unroll
  for (int j = 0; j < 8; ++j)
    if (bias_v[j] != 0.f) atomicAdd(&s_bias[c8 * 8 + j], bias_v[j]);
  __syncthreads();
  if (lr == 0) args.bias_part[l][(long)blockIdx.x * 64 + lc] = s_bias[lc];
}

__global__ void wgrad3_reduce_kernel(const Wg3Args args, int slabs,
                                     float* __restrict__ sq_part) {
  const int l = blockIdx.y;
  const int N = args.N[l], M = args.M[l];
  const float* __restrict__ part = args.part[l];
  const long NM = (long)N * M;
  const long waves = ((long)gridDim.x * blockDim.x) >> 6;
  const long wave_id = ((long)blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  const int lane = threadIdx.x & 63;
  float sq = 0.f;  // this block's sum of squared gradient elements
  for (long e = wave_id; e < NM; e += waves) {
    const int n = (int)(e / M), m = (int)(e % M);
    const long off = (long)n * 64 + m;
    float acc = 0.f;
    for (int s = lane; s < slabs; s += 64)
      acc += part[(long)s * (64 * 64) + off];
#pragma unroll
    for (int d = 32; d > 0; d >>= 1) acc += __shfl_down(acc, d);
    if (lane == 0) {
      args.dw[l][e] = acc;
      sq += acc * acc;
    }
  }
  for (long n = wave_id; n < N; n += waves) {
    float acc = 0.f;
    for (int s = lane; s < slabs; s += 64)
      acc += args.bias_part[l][(long)s * 64 + n];
#pragma unroll
    for (int d = 32; d > 0; d >>= 1) acc += __shfl_down(acc, d);
    if (lane == 0) {
      args.db[l][n] = acc;
      sq += acc * acc;
    }
  }
  if (sq_part != nullptr) {
    // block-reduce the per-thread sq partials (nonzero at lane 0s)
    __shared__ float smem[8];
    for (int off = 32; off > 0; off >>= 1) sq += __shfl_down(sq, off, 64);
    const int wv = threadIdx.x >> 6;
    if (lane == 0) smem[wv] = sq;
    __syncthreads();
    if (threadIdx.x == 0) {
      float t = 0.f;
      for (int w = 0; w < (int)(blockDim.x >> 6); ++w) t += smem[w];
      sq_part[(long)blockIdx.y * gridDim.x + blockIdx.x] = t;
    }
  }
}

// grad-clip coefficient straight from the wgrad reduce's sq partials:
// ONE tiny kernel instead of re-reading every gradient tensor.
__global__ void wgrad_clip_finalize_k(const float* __restrict__ sq_part,
                                      const int n_part,
                                      const float max_norm,
                                      float* __restrict__ coef,
                                      const int inverse) {
  float s = 0.f;
  for (int i = threadIdx.x; i < n_part; i += blockDim.x) s += sq_part[i];
  __shared__ float smem[8];
  for (int off = 32; off > 0; off >>= 1) s += __shfl_down(s, off, 64);
  const int lane = threadIdx.x & 63, wv = threadIdx.x >> 6;
  if (lane == 0) smem[wv] = s;
  __syncthreads();
  if (threadIdx.x == 0) {
    float t = 0.f;
    for (int w = 0; w < (int)(blockDim.x >> 6); ++w) t += smem[w];
    if (inverse) {
      const float c = (sqrtf(t) + 1e-6f) / max_norm;
      coef[0] = c > 1.f ? c : 1.f;
    } else {
      const float c = max_norm / (sqrtf(t) + 1e-6f);
      coef[0] = c < 1.f ? c : 1.f;
    }
  }
}

}  // namespace

extern "C" int wgrad3_slab_count_n(long K, int n_layers) {
  // target ~768 workgroups TOTAL across the layer axis: partials
  // traffic (written + re-read by the reduce) scales with slabs, so
  // more layers per launch means fewer slabs each
  long target = 768 / (n_layers > 0 ? n_layers : 1);
  long k_slab = (K + target - 1) / target;
  k_slab = ((k_slab + ROW_CHUNK - 1) / ROW_CHUNK) * ROW_CHUNK;
  if (k_slab < ROW_CHUNK) k_slab = ROW_CHUNK;
  return (int)((K + k_slab - 1) / k_slab);
}

extern "C" int wgrad3_slab_count(long K) { return wgrad3_slab_count_n(K, 3); }

extern "C" void launch_wgrad3(const void* const* dy, const void* const* x,
                              float* const* part, float* const* bias_part,
                              float* const* dw, float* const* db,
                              const int* N, const int* M, int n_layers,
                              long K, float* sq_part, void* stream) {
  Wg3Args a;
  for (int l = 0; l < n_layers; ++l) {
    a.dy[l] = (const __hip_bfloat16*)dy[l];
    a.x[l] = (const __hip_bfloat16*)x[l];
    a.part[l] = part[l];
    a.bias_part[l] = bias_part[l];
    a.dw[l] = dw[l];
    a.db[l] = db[l];
    a.N[l] = N[l];
    a.M[l] = M[l];
  }
  const Wg3Args* args = &a;
  long target = 768 / (n_layers > 0 ? n_layers : 1);
  long k_slab = (K + target - 1) / target;
  k_slab = ((k_slab + ROW_CHUNK - 1) / ROW_CHUNK) * ROW_CHUNK;
  if (k_slab < ROW_CHUNK) k_slab = ROW_CHUNK;
  const int slabs = (int)((K + k_slab - 1) / k_slab);
  hipLaunchKernelGGL(wgrad3_mfma_kernel, dim3(slabs, n_layers),
                     dim3(WG_THREADS), 0, (hipStream_t)stream, *args, K,
                     (int)k_slab);
  const int blocks = 512;  // n * 512 WGs cover the chip for the reduce
  hipLaunchKernelGGL(wgrad3_reduce_kernel, dim3(blocks, n_layers), dim3(256),
                     0, (hipStream_t)stream, *args, slabs, sq_part);
}

extern "C" void launch_wgrad_clip_finalize(const float* sq_part, int n_part,
                                           float max_norm, float* coef,
                                           int inverse, void* stream) {
  hipLaunchKernelGGL(wgrad_clip_finalize_k, dim3(1), dim3(256), 0,
                     (hipStream_t)stream, sq_part, n_part, max_norm, coef,
                     inverse);
}
