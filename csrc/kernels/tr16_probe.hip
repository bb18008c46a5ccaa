// Empirical probe of gfx950 `ds_read_b64_tr_b16` lane/element semantics.
//
// Fills LDS with bf16 value = element index, issues the transpose read with
// a parameterized per-lane address scheme, and dumps what each lane's four
// result elements contain. The derived mapping drives the flash-attention
// backward's B-fragment reads (replacing the b16 transpose-scatter image).
//
// Guide hypothesis (cdna_hip_programming.md §2): with the right address
// scheme, result[lane l][elem j] = lds[(l&15) + j*16 + (l>>4)*64] — i.e.
// column (l&15) of a [4][16] row-major bf16 block per 16-lane group.

#include <hip/hip_bf16.h>
#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstdio>

using bf16 = __hip_bfloat16;

// scheme 0: addr = base + l*8                  (lane-linear 8-B chunks)
// scheme 1: addr = base + (l&3)*32 + ((l>>2)&3)*8 + (l>>4)*128
//           (4x4-quad transpose hypothesis: lane supplies row (l&3),
//            col-quad ((l>>2)&3), 4x16-block (l>>4))
// scheme 2: addr = base + (l&15)*8 + (l>>4)*128 (row-of-8B per 16-lane grp)
__global__ void tr16_probe_kernel(float* out, int scheme) {
  __shared__ __align__(16) short lds[512];
  const int t = threadIdx.x;
  for (int i = t; i < 512; i += blockDim.x) {
    lds[i] = (short)i;  // value = element index (bf16 bit pattern abuse is
                        // fine: we only compare raw shorts)
  }
  __syncthreads();
  if (t >= 64) return;
  const int l = t;
  int addr;
  switch (scheme) {
    case 1: addr = (l & 3) * 32 + ((l >> 2) & 3) * 8 + (l >> 4) * 128; break;
    case 2: addr = (l & 15) * 8 + (l >> 4) * 128; break;
    default: addr = l * 8; break;
  }
  // byte address into LDS
  uint32_t v2[2];
  asm volatile("ds_read_b64_tr_b16 %0, %2 offset:0\n\ts_waitcnt lgkmcnt(0)"
               : "=v"(*(unsigned long long*)v2)
               : "v"((unsigned)(addr + (unsigned)(uintptr_t)&lds[0])), "v"(addr));
  const short* r = (const short*)v2;
  for (int j = 0; j < 4; j++) out[l * 4 + j] = (float)r[j];
}

extern "C" int run_tr16_probe() {
  float* d;
  (void)hipMalloc(&d, 64 * 4 * sizeof(float));
  float h[256];
  for (int scheme = 0; scheme < 3; scheme++) {
    hipLaunchKernelGGL(tr16_probe_kernel, dim3(1), dim3(64), 0, 0, d, scheme);
    (void)hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
    printf("== scheme %d: result[lane][elem] = lds element index ==\n", scheme);
    for (int l = 0; l < 64; l++) {
      printf("l%02d: %4.0f %4.0f %4.0f %4.0f%s", l, h[l * 4], h[l * 4 + 1],
             h[l * 4 + 2], h[l * 4 + 3], (l % 4 == 3) ? "\n" : "   ");
    }
    // check the guide formula: result[l][j] == (l&15) + j*16 + (l>>4)*64
    int ok = 1;
    for (int l = 0; l < 64 && ok; l++)
      for (int j = 0; j < 4; j++)
        if ((int)h[l * 4 + j] != ((l & 15) + j * 16 + (l >> 4) * 64)) ok = 0;
    printf("scheme %d matches guide formula: %s\n", scheme, ok ? "YES" : "no");
  }
  (void)hipFree(d);
  return 0;
}

int main() { return run_tr16_probe(); }
