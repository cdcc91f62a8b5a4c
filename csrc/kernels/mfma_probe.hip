// MFMA fragment-layout probe for v_mfma_f32_32x32x16_bf16 on gfx950.
// Computes D = A(32x16) x B(16x32) with the hypothesized lane->element
// maps and writes D to global memory; the host compares against a CPU
// reference (asymmetric A and B so transposes are caught).
//
// Hypothesized layouts (CDNA3 32x32x8 pattern with K doubled):
//   A[m][k]: lane l holds m = l%32, k = (l/32)*8 + j   (j = 0..7)
//   B[k][n]: lane l holds n = l%32, k = (l/32)*8 + j
//   C/D    : col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
//            (authoritative, cdna_hip_programming.md section 3)

#include <hip/hip_runtime.h>

using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x16 = __attribute__((ext_vector_type(16))) float;

__global__ void mfma_probe_kernel(const unsigned short* __restrict__ A,
                                  const unsigned short* __restrict__ B,
                                  float* __restrict__ D) {
  const int lane = threadIdx.x & 63;
  const int half = lane >> 5;     // 0 or 1
  const int lm = lane & 31;
  bf16x8 a, b;
  for (int j = 0; j < 8; ++j) {
    // A is 32x16 row-major; B is 16x32 row-major
    a[j] = (short)A[lm * 16 + (half * 8 + j)];
    b[j] = (short)B[(half * 8 + j) * 32 + lm];
  }
  f32x16 c = {};
  c = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
  for (int reg = 0; reg < 16; ++reg) {
    const int row = (reg & 3) + 8 * (reg >> 2) + 4 * half;
    const int col = lm;
    D[row * 32 + col] = c[reg];
  }
}

extern "C" void run_mfma_probe(const unsigned short* A,
                               const unsigned short* B, float* D,
                               hipStream_t stream) {
  hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, stream, A, B,
                     D);
}
