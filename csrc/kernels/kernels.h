// Shared launcher declarations between the .hip kernel TUs and the
// torch-extension bindings TU. Streams are passed as the raw HIP stream
// pointer (forward-declared so the bindings TU needs no HIP headers).
#pragma once

#include <cstdint>

struct ihipStream_t;
using tft_stream = ihipStream_t*;

namespace torchft_amd {

// fp8_quant.hip ------------------------------------------------------------
// dtype_code: 0 = bf16, 1 = fp16, 2 = fp32
void launch_quantize_dtype(int dtype_code, const int64_t* ptrs,
                           const int64_t* block_prefix, const int64_t* numels,
                           int n_tensors, int64_t total_blocks,
                           int64_t padded_blocks, int64_t blocks_per_rank,
                           int64_t slice_bytes, uint8_t* pack, tft_stream stream);

void launch_dequantize_dtype(int dtype_code, const int64_t* ptrs,
                             const int64_t* block_prefix, const int64_t* numels,
                             int n_tensors, int64_t total_blocks,
                             int64_t padded_blocks, int64_t blocks_per_rank,
                             int64_t slice_bytes, const uint8_t* pack,
                             tft_stream stream);

void launch_reduce(const uint8_t* in, uint8_t* out, int world,
                   int64_t blocks_per_rank, int64_t slice_bytes, bool avg,
                   tft_stream stream);

// fused_model_ops.hip ------------------------------------------------------
void launch_rmsnorm_fwd(const void* x, const void* w, void* y, float* invrms,
                        int64_t rows, int H, float eps, tft_stream stream);
void launch_rmsnorm_bwd(const void* dy, const void* x, const void* w,
                        const float* invrms, void* dx, float* dw, int64_t rows,
                        int H, tft_stream stream);
void launch_rope(const void* x, void* out, const float* cos_tab,
                 const float* sin_tab, int64_t total_pairs, int S, int Hh, int D,
                 bool backward, tft_stream stream);
void launch_swiglu_fwd(const void* a, const void* b, void* out, int64_t n,
                       tft_stream stream);
void launch_swiglu_glu_fwd(const void* gu, void* out, int64_t rows, int64_t f,
                           tft_stream stream);
void launch_swiglu_glu_bwd(const void* dy, const void* gu, void* dgu,
                           int64_t rows, int64_t f, tft_stream stream);
void launch_swiglu_bwd(const void* dy, const void* a, const void* b, void* da,
                       void* db, int64_t n, tft_stream stream);

// fused_adamw.hip ----------------------------------------------------------
void launch_adamw(const int64_t* param_ptrs, const int64_t* grad_ptrs,
                  const int64_t* m_ptrs, const int64_t* v_ptrs,
                  const int64_t* numels, const int64_t* block_prefix,
                  int n_tensors, int64_t total_blocks, float lr, float beta1,
                  float beta2, float eps, float weight_decay,
                  float bias_correction1, float bias_correction2,
                  tft_stream stream);

}  // namespace torchft_amd

namespace torchft_amd {

// flash_attn_bwd.hip --------------------------------------------------------
void launch_fa_fwd(const void* q, const void* k, const void* v, void* out,
                   float* lse, int B, int Hq, int Hkv, int S, float scale,
                   bool causal, bool bshd, tft_stream stream);
void launch_fa_delta(const void* dout, const void* o, float* delta,
                     int64_t rows, int Hq, int S, bool bshd, tft_stream stream);
void launch_fa_bwd(const void* q, const void* k, const void* v, const void* dout,
                   const float* lse, const float* delta, void* dq, void* dk,
                   void* dv, int B, int Hq, int Hkv, int S, float scale,
                   bool causal, bool bshd, tft_stream stream);

// mfma_probe.hip ------------------------------------------------------------
void launch_mfma_probe(const void* A, const void* B, float* D, tft_stream s);

}  // namespace torchft_amd
