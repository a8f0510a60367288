// MFMA fragment-layout probe for gfx950 mfma_f32_16x16x32_bf16.
//
// For each (lane, reg) of the A fragment we set exactly that one input
// element to 1 (all else 0), multiply by a known B, and decode which
// (m, k) the element mapped to from the output row/col pattern.  Same
// for B.  C/D layout is known (guide): col = lane & 15,
// row = (lane >> 4) * 4 + reg.
//
// Build: hipcc --offload-arch=gfx950 -o mfma_probe mfma_probe.hip
// Run:   ./mfma_probe   (prints A and B lane->(m,k) tables)

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdio>

using frag_ab = __attribute__((ext_vector_type(8))) short;
using frag_cd = __attribute__((ext_vector_type(4))) float;

__device__ __forceinline__ short f2bf(float f) {
  union { float f; unsigned u; } v{f};
  return (short)(v.u >> 16);
}

// Probe A: for (sel_lane, sel_reg), a_frag has a single 1; B[k][n] = k + n/100.
// D[m][n] = A[m][k*] * B[k*][n] = B[k*][n] when m == m*; so the nonzero row
// of D is m*, and its value at n=0 is k*.
__global__ void probe_a_kernel(float* out /* [64*8, 2] -> (m, k) */) {
  const int lane = threadIdx.x;  // 64 lanes
  for (int sel_lane = 0; sel_lane < 64; ++sel_lane) {
    for (int sel_reg = 0; sel_reg < 8; ++sel_reg) {
      frag_ab a = {};
      frag_ab b;
      if (lane == sel_lane) a[sel_reg] = f2bf(1.0f);
      // B fragment: value = k + n/100 — but we must know B's layout to
      // fill it... chicken-and-egg. Instead fill EVERY b element with
      // its (lane, reg) code and decode k from the D VALUE below using
      // the B probe first. For the A probe we set B = all-ones: then
      // D[m*][n] = 1 for all n, revealing only m*. k comes from pass 2.
      for (int i = 0; i < 8; ++i) b[i] = f2bf(1.0f);
      frag_cd acc = {};
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
      // D: col = lane & 15, row = (lane>>4)*4 + reg
      for (int reg = 0; reg < 4; ++reg) {
        const int m = (lane >> 4) * 4 + reg;
        const int n = lane & 15;
        if (acc[reg] != 0.0f && n == 0) {
          out[(sel_lane * 8 + sel_reg) * 2 + 0] = (float)m;
        }
      }
      __syncthreads();
    }
  }
}

// Probe pass 2 for A's k index: set a single A element (sel) to 1 AND a
// single B element (probe every b (lane,reg)) to 1: D nonzero iff the k
// indices MATCH. Fix one a-sel, sweep b-sel: the matching b element's k
// equals a's k. But B's layout is also unknown — so instead probe k by
// contracting with B filled as b_value = k directly impossible without
// layout...  Pragmatic approach: assume the standard CDNA mapping
// families and TEST them: candidate k(lane, reg) ∈ {8*(lane>>4)+reg,
// 4*(lane>>4)+reg + 16*(reg>=4), reg*4 + ... }. We verify end-to-end:
// fill A[m][k] = m*32 + k and B[k][n] = (k==n) ? 1 : 0 under a candidate
// layout and check D[m][n] == A[m][n].
struct Cand { int id; };

__device__ int k_of(int cand, int lane, int reg) {
  switch (cand) {
    case 0: return 8 * (lane >> 4) + reg;                      // contiguous 8
    case 1: return 4 * (lane >> 4) + reg % 4 + 16 * (reg / 4); // split halves
    case 2: return (lane >> 4) + 4 * reg;                      // interleaved
    default: return 0;
  }
}

__global__ void verify_cand_kernel(int cand, float* max_err) {
  const int lane = threadIdx.x;
  // A[m][k] = m + k * 0.01 ; B[k][n] = (k % 16 == n) ? 1+k/100.0 : 0
  // D[m][n] = sum_k A[m][k] * B[k][n] = A[m][n]*(1+n/100) + A[m][n+16]*(1+(n+16)/100)
  frag_ab a, b;
  const int am = lane & 15;
  const int bn = lane & 15;
  for (int reg = 0; reg < 8; ++reg) {
    const int ak = k_of(cand, lane, reg);
    a[reg] = f2bf((float)am + 0.01f * (float)ak);
    const int bk = k_of(cand, lane, reg);
    b[reg] = f2bf((bk % 16 == bn) ? (1.0f + (float)bk / 100.0f) : 0.0f);
  }
  frag_cd acc = {};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
  float err = 0.f;
  for (int reg = 0; reg < 4; ++reg) {
    const int m = (lane >> 4) * 4 + reg;
    const int n = lane & 15;
    // expected with bf16 rounding tolerance
    auto bf = [](float f) { union { float f; unsigned u; } v{f}; v.u &= 0xFFFF0000u; return v.f; };
    const float a1 = bf((float)m + 0.01f * (float)n);
    const float a2 = bf((float)m + 0.01f * (float)(n + 16));
    const float w1 = bf(1.0f + (float)n / 100.0f);
    const float w2 = bf(1.0f + (float)(n + 16) / 100.0f);
    const float expect = a1 * w1 + a2 * w2;
    err = fmaxf(err, fabsf(acc[reg] - expect));
  }
  atomicMax((int*)max_err, __float_as_int(err));
}

// ---- 32x32x16 bf16 verification (same method) -----------------------------
// Candidate mappings: A row = lane&31, B col = lane&31,
// k = 8*(lane>>5) + reg (8 regs).  C/D (guide, hardware-verified m74/m101):
// col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5), reg in [0,16).
using frag_cd32 = __attribute__((ext_vector_type(16))) float;

__global__ void verify32_kernel(float* max_err) {
  const int lane = threadIdx.x;
  frag_ab a, b;
  const int am = lane & 31;
  const int bn = lane & 31;
  for (int reg = 0; reg < 8; ++reg) {
    const int k = 8 * (lane >> 5) + reg;
    a[reg] = f2bf((float)am + 0.01f * (float)k);
    b[reg] = f2bf((k == bn % 16) ? (1.0f + (float)k / 100.0f) : 0.0f);
  }
  frag_cd32 acc = {};
  acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc, 0, 0, 0);
  auto bf = [](float f) { union { float f; unsigned u; } v{f}; v.u &= 0xFFFF0000u; return v.f; };
  float err = 0.f;
  for (int reg = 0; reg < 16; ++reg) {
    const int m = (reg & 3) + 8 * (reg >> 2) + 4 * (lane >> 5);
    const int n = lane & 31;
    // expected: D[m][n] = A[m][n%16] * (1 + (n%16)/100)
    const float a1 = bf((float)m + 0.01f * (float)(n % 16));
    const float w1 = bf(1.0f + (float)(n % 16) / 100.0f);
    const float expect = a1 * w1;
    err = fmaxf(err, fabsf(acc[reg] - expect));
  }
  atomicMax((int*)max_err, __float_as_int(err));
}



int main() {
  float* d_err;
  hipMalloc(&d_err, sizeof(float));
  for (int cand = 0; cand < 3; ++cand) {
    float zero = 0.f;
    hipMemcpy(d_err, &zero, sizeof(float), hipMemcpyHostToDevice);
    hipLaunchKernelGGL(verify_cand_kernel, dim3(1), dim3(64), 0, 0, cand, d_err);
    hipDeviceSynchronize();
    float err;
    hipMemcpy(&err, d_err, sizeof(float), hipMemcpyDeviceToHost);
    printf("candidate %d: max |err| = %f  %s\n", cand, err,
           err < 0.15f ? "<-- MATCHES" : "");
  }
  {
    float zero = 0.f;
    hipMemcpy(d_err, &zero, sizeof(float), hipMemcpyHostToDevice);
    hipLaunchKernelGGL(verify32_kernel, dim3(1), dim3(64), 0, 0, d_err);
    hipDeviceSynchronize();
    float err;
    hipMemcpy(&err, d_err, sizeof(float), hipMemcpyDeviceToHost);
    printf("32x32x16 candidate: max |err| = %f  %s\n", err,
           err < 0.15f ? "<-- MATCHES" : "");
  }
  return 0;
}

