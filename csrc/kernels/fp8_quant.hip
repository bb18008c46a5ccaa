// Fused OCP-fp8(e4m3) block quantize / dequantize / reduce kernels for the
// quantized gradient collectives — CDNA4 (gfx950) native.
//
// Capability parity with the reference Triton trio
// (torchft/quantization.py:53-428) re-designed for MI355X:
//  * fixed 2048-element blocks (wave-64-friendly; uniform grid) instead of
//    per-row striping — one fp32 scale per block, payload stays 4B-aligned
//  * per-tensor block padding so a block never straddles tensors: the
//    block→tensor lookup is a wave-uniform scalar binary search
//  * bf16/fp16/fp32 inputs load 16 B/lane (vectorized); fp8 conversion uses
//    the native OCP e4m3 type (gfx950 v_cvt path) — NOT MI300X fnuz
//  * accumulation in the reduce kernel runs in fp32 in fixed rank order
//    0..W-1 so every rank computes bitwise-identical reductions
//
// Packed wire layout (what goes over RCCL alltoall/allgather, all-7-xGMI
// friendly): W equal slices, slice r = [fp32 dequant-scales of its blocks]
// ‖ [2048 fp8 bytes per block]. slice_bytes = B_pr*(4+2048).

#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <hip/hip_fp8.h>
#include <hip/hip_runtime.h>

#include <cstdint>

#define FP8_MAX 448.0f
#define QBLOCK 2048          // elements per quantization block
#define QTHREADS 256         // threads per workgroup
#define ELEMS_PER_THREAD 8   // QBLOCK / QTHREADS

namespace torchft_amd {

using fp8_t = __hip_fp8_e4m3;  // OCP e4m3fn (gfx950); NOT the MI300X fnuz type

// scalar conversions (torch extensions build with __HIP_NO_HALF_CONVERSIONS__)
__device__ inline float to_f32(__hip_bfloat16 v) { return __bfloat162float(v); }
__device__ inline float to_f32(__half v) { return __half2float(v); }
__device__ inline float to_f32(float v) { return v; }
__device__ inline void from_f32(__hip_bfloat16* d, float v) { *d = __float2bfloat16(v); }
__device__ inline void from_f32(__half* d, float v) { *d = __float2half(v); }
__device__ inline void from_f32(float* d, float v) { *d = v; }

// ---- vector load helpers: 16B/lane for bf16/fp16, 2x16B for fp32 ----------

template <typename T>
struct VecIO;

template <>
struct VecIO<__hip_bfloat16> {
  // 8 bf16 = 16 bytes
  static __device__ inline void load8(const __hip_bfloat16* p, float (&v)[8]) {
    const uint4 raw = *reinterpret_cast<const uint4*>(p);  // 16 bytes
    const __hip_bfloat162* h = reinterpret_cast<const __hip_bfloat162*>(&raw);
#pragma unroll
    for (int i = 0; i < 4; i++) {
      float2 f = __bfloat1622float2(h[i]);
      v[2 * i] = f.x;
      v[2 * i + 1] = f.y;
    }
  }
  static __device__ inline void store8(__hip_bfloat16* p, const float (&v)[8]) {
    uint4 raw;
    __hip_bfloat162* h = reinterpret_cast<__hip_bfloat162*>(&raw);
#pragma unroll
    for (int i = 0; i < 4; i++) {
      h[i] = __float22bfloat162_rn(make_float2(v[2 * i], v[2 * i + 1]));
    }
    *reinterpret_cast<uint4*>(p) = raw;
  }
};

template <>
struct VecIO<__half> {
  static __device__ inline void load8(const __half* p, float (&v)[8]) {
    const uint4 raw = *reinterpret_cast<const uint4*>(p);  // 16 bytes
    const __half2* h = reinterpret_cast<const __half2*>(&raw);
#pragma unroll
    for (int i = 0; i < 4; i++) {
      float2 f = __half22float2(h[i]);
      v[2 * i] = f.x;
      v[2 * i + 1] = f.y;
    }
  }
  static __device__ inline void store8(__half* p, const float (&v)[8]) {
    uint4 raw;
    __half2* h = reinterpret_cast<__half2*>(&raw);
#pragma unroll
    for (int i = 0; i < 4; i++) {
      h[i] = __float22half2_rn(make_float2(v[2 * i], v[2 * i + 1]));
    }
    *reinterpret_cast<uint4*>(p) = raw;
  }
};

template <>
struct VecIO<float> {
  static __device__ inline void load8(const float* p, float (&v)[8]) {
    const float4 a = *reinterpret_cast<const float4*>(p);
    const float4 b = *reinterpret_cast<const float4*>(p + 4);
    v[0] = a.x; v[1] = a.y; v[2] = a.z; v[3] = a.w;
    v[4] = b.x; v[5] = b.y; v[6] = b.z; v[7] = b.w;
  }
  static __device__ inline void store8(float* p, const float (&v)[8]) {
    *reinterpret_cast<float4*>(p) = make_float4(v[0], v[1], v[2], v[3]);
    *reinterpret_cast<float4*>(p + 4) = make_float4(v[4], v[5], v[6], v[7]);
  }
};

// ---- block reduce (max) across the 4 waves of a 256-thread workgroup ------

__device__ inline float block_reduce_max(float v) {
  // wave64 reduce
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    v = fmaxf(v, __shfl_down(v, off, 64));
  }
  __shared__ float warp_max[QTHREADS / 64];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  if (lane == 0) warp_max[wave] = v;
  __syncthreads();
  float m = warp_max[0];
#pragma unroll
  for (int w = 1; w < QTHREADS / 64; w++) m = fmaxf(m, warp_max[w]);
  return m;  // every thread gets the block max
}

// ---- block→tensor lookup: wave-uniform scalar binary search ---------------

// block_prefix has n_tensors+1 entries: tensor t owns blocks
// [block_prefix[t], block_prefix[t+1])
__device__ inline int find_tensor(const int64_t* block_prefix, int n_tensors,
                                  int64_t block) {
  int lo = 0, hi = n_tensors;
  while (hi - lo > 1) {
    int mid = (lo + hi) >> 1;
    if (block_prefix[mid] <= block)
      lo = mid;
    else
      hi = mid;
  }
  return lo;
}

// ---- quantize --------------------------------------------------------------

// Grid: one workgroup per (padded) block. Blocks >= total_blocks write
// scale=0 + zero payload (padding slots).
template <typename T>
__global__ void quantize_fp8_kernel(
    const int64_t* __restrict__ tensor_ptrs,   // [n] device addresses
    const int64_t* __restrict__ block_prefix,  // [n+1]
    const int64_t* __restrict__ numels,        // [n]
    int n_tensors, int64_t total_blocks, int64_t blocks_per_rank,
    int64_t slice_bytes, uint8_t* __restrict__ pack) {
  const int64_t b = blockIdx.x;
  // destination inside the packed wire buffer
  const int64_t r = b / blocks_per_rank;
  const int64_t bl = b % blocks_per_rank;
  uint8_t* slice = pack + r * slice_bytes;
  float* scale_out = reinterpret_cast<float*>(slice) + bl;
  uint8_t* payload = slice + blocks_per_rank * 4 + bl * QBLOCK;

  float v[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  bool valid = b < total_blocks;
  int64_t offset_in_t = 0;
  const T* src = nullptr;
  int64_t numel = 0;
  if (valid) {
    const int t = find_tensor(block_prefix, n_tensors, b);
    offset_in_t = (b - block_prefix[t]) * QBLOCK + threadIdx.x * ELEMS_PER_THREAD;
    src = reinterpret_cast<const T*>(tensor_ptrs[t]);
    numel = numels[t];
    if (offset_in_t + ELEMS_PER_THREAD <= numel) {
      VecIO<T>::load8(src + offset_in_t, v);
    } else {
#pragma unroll
      for (int i = 0; i < ELEMS_PER_THREAD; i++) {
        v[i] = (offset_in_t + i < numel) ? to_f32(src[offset_in_t + i]) : 0.0f;
      }
    }
  }

  float amax = 0.f;
#pragma unroll
  for (int i = 0; i < ELEMS_PER_THREAD; i++) amax = fmaxf(amax, fabsf(v[i]));
  amax = block_reduce_max(amax);

  const float q_scale = amax > 0.f ? (FP8_MAX / amax) : 0.f;
  const float dq_scale = amax > 0.f ? (amax / FP8_MAX) : 0.f;

  uint8_t q[8];
#pragma unroll
  for (int i = 0; i < ELEMS_PER_THREAD; i++) {
    fp8_t f8(v[i] * q_scale);
    q[i] = *reinterpret_cast<uint8_t*>(&f8);
  }
  *reinterpret_cast<uint2*>(payload + threadIdx.x * ELEMS_PER_THREAD) =
      *reinterpret_cast<uint2*>(q);
  if (threadIdx.x == 0) *scale_out = dq_scale;
}

// ---- dequantize back into the tensors --------------------------------------

template <typename T>
__global__ void dequantize_fp8_kernel(
    const int64_t* __restrict__ tensor_ptrs,
    const int64_t* __restrict__ block_prefix,
    const int64_t* __restrict__ numels,
    int n_tensors, int64_t total_blocks, int64_t blocks_per_rank,
    int64_t slice_bytes, const uint8_t* __restrict__ pack) {
  const int64_t b = blockIdx.x;
  if (b >= total_blocks) return;  // padding slots map to no tensor
  const int64_t r = b / blocks_per_rank;
  const int64_t bl = b % blocks_per_rank;
  const uint8_t* slice = pack + r * slice_bytes;
  const float dq_scale = reinterpret_cast<const float*>(slice)[bl];
  const uint8_t* payload = slice + blocks_per_rank * 4 + bl * QBLOCK;

  const int t = find_tensor(block_prefix, n_tensors, b);
  const int64_t offset_in_t =
      (b - block_prefix[t]) * QBLOCK + threadIdx.x * ELEMS_PER_THREAD;
  T* dst = reinterpret_cast<T*>(tensor_ptrs[t]);
  const int64_t numel = numels[t];

  uint8_t q[8];
  *reinterpret_cast<uint2*>(q) =
      *reinterpret_cast<const uint2*>(payload + threadIdx.x * ELEMS_PER_THREAD);
  float v[8];
#pragma unroll
  for (int i = 0; i < ELEMS_PER_THREAD; i++) {
    fp8_t f8 = *reinterpret_cast<fp8_t*>(&q[i]);
    v[i] = (float)f8 * dq_scale;
  }
  if (offset_in_t + ELEMS_PER_THREAD <= numel) {
    VecIO<T>::store8(dst + offset_in_t, v);
  } else {
    for (int i = 0; i < ELEMS_PER_THREAD; i++) {
      if (offset_in_t + i < numel) from_f32(&dst[offset_in_t + i], v[i]);
    }
  }
}

// ---- reduce W copies of this rank's slice ----------------------------------

// in: [world, slice_bytes] (every rank's quantized copy of OUR blocks),
// out: [slice_bytes] requantized sum (or avg). Accumulates fp32 in fixed
// rank order 0..W-1 → bitwise deterministic regardless of which rank runs it.
__global__ void reduce_fp8_slices_kernel(const uint8_t* __restrict__ in,
                                         uint8_t* __restrict__ out, int world,
                                         int64_t blocks_per_rank,
                                         int64_t slice_bytes, float inv_div) {
  const int64_t bl = blockIdx.x;  // local block index within the slice
  float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};

  for (int r = 0; r < world; r++) {
    const uint8_t* slice = in + (int64_t)r * slice_bytes;
    const float dq_scale = reinterpret_cast<const float*>(slice)[bl];
    const uint8_t* payload = slice + blocks_per_rank * 4 + bl * QBLOCK;
    uint8_t q[8];
    *reinterpret_cast<uint2*>(q) =
        *reinterpret_cast<const uint2*>(payload + threadIdx.x * ELEMS_PER_THREAD);
#pragma unroll
    for (int i = 0; i < ELEMS_PER_THREAD; i++) {
      fp8_t f8 = *reinterpret_cast<fp8_t*>(&q[i]);
      acc[i] += (float)f8 * dq_scale;
    }
  }

  float amax = 0.f;
#pragma unroll
  for (int i = 0; i < ELEMS_PER_THREAD; i++) amax = fmaxf(amax, fabsf(acc[i]));
  amax = block_reduce_max(amax);

  const float q_scale = amax > 0.f ? (FP8_MAX / amax) : 0.f;
  // inv_div folds AVG's 1/world into the stored dequant scale — the payload
  // bytes are identical for SUM and AVG.
  const float dq_scale = amax > 0.f ? (amax / FP8_MAX) * inv_div : 0.f;

  float* scale_out = reinterpret_cast<float*>(out) + bl;
  uint8_t* payload_out = out + blocks_per_rank * 4 + bl * QBLOCK;
  uint8_t q[8];
#pragma unroll
  for (int i = 0; i < ELEMS_PER_THREAD; i++) {
    fp8_t f8(acc[i] * q_scale);
    q[i] = *reinterpret_cast<uint8_t*>(&f8);
  }
  *reinterpret_cast<uint2*>(payload_out + threadIdx.x * ELEMS_PER_THREAD) =
      *reinterpret_cast<uint2*>(q);
  if (threadIdx.x == 0) *scale_out = dq_scale;
}

// ---- host-visible launchers (raw pointers; stream passed in) ---------------

template <typename T>
void launch_quantize(const int64_t* ptrs, const int64_t* block_prefix,
                     const int64_t* numels, int n_tensors, int64_t total_blocks,
                     int64_t padded_blocks, int64_t blocks_per_rank,
                     int64_t slice_bytes, uint8_t* pack, hipStream_t stream) {
  hipLaunchKernelGGL(quantize_fp8_kernel<T>, dim3((uint32_t)padded_blocks),
                     dim3(QTHREADS), 0, stream, ptrs, block_prefix, numels,
                     n_tensors, total_blocks, blocks_per_rank, slice_bytes, pack);
}

template <typename T>
void launch_dequantize(const int64_t* ptrs, const int64_t* block_prefix,
                       const int64_t* numels, int n_tensors, int64_t total_blocks,
                       int64_t padded_blocks, int64_t blocks_per_rank,
                       int64_t slice_bytes, const uint8_t* pack,
                       hipStream_t stream) {
  hipLaunchKernelGGL(dequantize_fp8_kernel<T>, dim3((uint32_t)padded_blocks),
                     dim3(QTHREADS), 0, stream, ptrs, block_prefix, numels,
                     n_tensors, total_blocks, blocks_per_rank, slice_bytes, pack);
}

void launch_reduce(const uint8_t* in, uint8_t* out, int world,
                   int64_t blocks_per_rank, int64_t slice_bytes, bool avg,
                   hipStream_t stream) {
  hipLaunchKernelGGL(reduce_fp8_slices_kernel, dim3((uint32_t)blocks_per_rank),
                     dim3(QTHREADS), 0, stream, in, out, world, blocks_per_rank,
                     slice_bytes, avg ? 1.0f / world : 1.0f);
}

// dtype_code: 0 = bf16, 1 = fp16, 2 = fp32 (matches kernels.h)
void launch_quantize_dtype(int dtype_code, const int64_t* ptrs,
                           const int64_t* block_prefix, const int64_t* numels,
                           int n_tensors, int64_t total_blocks,
                           int64_t padded_blocks, int64_t blocks_per_rank,
                           int64_t slice_bytes, uint8_t* pack, hipStream_t stream) {
  switch (dtype_code) {
    case 0:
      launch_quantize<__hip_bfloat16>(ptrs, block_prefix, numels, n_tensors,
                                      total_blocks, padded_blocks, blocks_per_rank,
                                      slice_bytes, pack, stream);
      break;
    case 1:
      launch_quantize<__half>(ptrs, block_prefix, numels, n_tensors, total_blocks,
                              padded_blocks, blocks_per_rank, slice_bytes, pack,
                              stream);
      break;
    default:
      launch_quantize<float>(ptrs, block_prefix, numels, n_tensors, total_blocks,
                             padded_blocks, blocks_per_rank, slice_bytes, pack,
                             stream);
  }
}

void launch_dequantize_dtype(int dtype_code, const int64_t* ptrs,
                             const int64_t* block_prefix, const int64_t* numels,
                             int n_tensors, int64_t total_blocks,
                             int64_t padded_blocks, int64_t blocks_per_rank,
                             int64_t slice_bytes, const uint8_t* pack,
                             hipStream_t stream) {
  switch (dtype_code) {
    case 0:
      launch_dequantize<__hip_bfloat16>(ptrs, block_prefix, numels, n_tensors,
                                        total_blocks, padded_blocks, blocks_per_rank,
                                        slice_bytes, pack, stream);
      break;
    case 1:
      launch_dequantize<__half>(ptrs, block_prefix, numels, n_tensors, total_blocks,
                                padded_blocks, blocks_per_rank, slice_bytes, pack,
                                stream);
      break;
    default:
      launch_dequantize<float>(ptrs, block_prefix, numels, n_tensors, total_blocks,
                               padded_blocks, blocks_per_rank, slice_bytes, pack,
                               stream);
  }
}

}  // namespace torchft_amd
