// MFMA layout probe for mfma_f32_32x32x16_bf16 (gfx950).
//
// Empirically verifies the lane→element mappings this codebase assumes:
//   A (8 bf16/lane): A[row = lane&31][k = (lane>>5)*8 + m], m in 0..7
//   B (8 bf16/lane): B[k = (lane>>5)*8 + m][col = lane&31]
//   C/D (16 f32/lane): col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
// The probe computes D = A·B for A,B filled with position-encoded values and
// writes D to global following the assumed C layout; the host checks it
// against a plain matmul. Run once per toolchain bump.

#include <hip/hip_bf16.h>
#include <hip/hip_runtime.h>

namespace torchft_amd {

using bf16 = __hip_bfloat16;
typedef __attribute__((__vector_size__(8 * sizeof(short)))) short bf16x8_vec;
typedef __attribute__((__vector_size__(16 * sizeof(float)))) float f32x16;

__global__ void mfma_probe_kernel(const bf16* __restrict__ A,  // [32][16]
                                  const bf16* __restrict__ B,  // [16][32]
                                  float* __restrict__ D) {     // [32][32]
  const int lane = threadIdx.x & 63;
  bf16x8_vec a_frag, b_frag;
  const int arow = lane & 31;
  const int kbase = (lane >> 5) * 8;
#pragma unroll
  for (int m = 0; m < 8; m++) {
    reinterpret_cast<bf16*>(&a_frag)[m] = A[arow * 16 + kbase + m];
    reinterpret_cast<bf16*>(&b_frag)[m] = B[(kbase + m) * 32 + (lane & 31)];
  }
  f32x16 acc = {};
  acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a_frag, b_frag, acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 16; r++) {
    const int row = (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
    const int col = lane & 31;
    D[row * 32 + col] = acc[r];
  }
}

void launch_mfma_probe(const void* A, const void* B, float* D, hipStream_t s) {
  hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, s,
                     (const bf16*)A, (const bf16*)B, D);
}

}  // namespace torchft_amd
