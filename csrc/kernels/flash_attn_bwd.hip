// Flash-attention BACKWARD for CDNA4 (gfx950) — bf16, D=128, causal + full,
// GQA-native (dK/dV accumulate over the query heads of each KV head in
// registers; no atomics anywhere).
//
// Motivation (profiles/SUMMARY.md): the stock aotriton backward is 31% of
// the Llama-8B step; this is a hand-written MFMA replacement. The forward
// stays on aten's flash kernel — we consume its logsumexp
// (P = exp(scale*S - LSE)) plus the usual FA2 delta D_i = rowsum(dO*O).
//
// Structure (FA2 split): kernel 1 computes dK/dV (grid over KV blocks),
// kernel 2 computes dQ (grid over Q blocks); both recompute S/P per tile.
// Each workgroup = 4 wave64s; each wave owns one 32-row block and holds its
// fp32 accumulators in the unified VGPR/AGPR file.
//
// Schedule (v2): double-buffered LDS staging pipelined T14-style — next
// tile's global loads issue an iteration early, the ds_write pass lands in
// the buffer the previous iteration finished reading, ONE barrier per
// iteration; the S and dP MFMA chains run interleaved (independent
// accumulators hide the 32x32 dependent-accumulator latency), and P/dS use
// separate per-wave transpose buffers so all 16 dV+dK MFMAs interleave.
//
// MFMA lane mappings (verified by mfma_probe.hip on gfx950):
//   A: row=lane&31, k=(lane>>5)*8+m   B: k=(lane>>5)*8+m, col=lane&31
//   C/D: col=lane&31, row=(reg&3)+8*(reg>>2)+4*(lane>>5)

#include <hip/hip_bf16.h>
#include <hip/hip_runtime.h>

#include <cstdint>

namespace torchft_amd {

using bf16 = __hip_bfloat16;
typedef __attribute__((__vector_size__(8 * sizeof(short)))) short bf16x8_vec;
typedef __attribute__((__vector_size__(16 * sizeof(float)))) float f32x16;

#define FA_D 128
#define FA_BLK 32
#define FA_WAVES 4
#define FA_THREADS 256

// ---- LDS image geometry ---------------------------------------------------
__device__ inline int rm_addr(int row, int byte_off) {
  return row * 256 + (byte_off ^ ((row & 15) << 4));  // 32 x 256 B, XOR-16
}
// 64-B rows alias a 256-B LDS bank row every 4 rows, and the b16 transpose
// scatter writes 16 rows that are congruent mod 16 — so the XOR mixes BOTH
// (row>>2) (de-conflicts the 16-consecutive-row b128 reads) and (row>>4)
// (spreads the stride-16 write pattern): measured 4-way reads / 16-way
// writes with a single-term XOR.
__device__ inline int row_swz(int row, int byte_off) {
  return byte_off ^ ((((row >> 2) ^ (row >> 4)) & 3) << 4);
}
__device__ inline int tr_addr(int d, int byte_off) {
  return d * 64 + row_swz(d, byte_off);  // 128 x 64 B
}
__device__ inline int pb_addr(int row, int byte_off) {
  return row * 64 + row_swz(row, byte_off);  // 32 x 64 B
}

struct ImageSet {
  __align__(16) unsigned char a_rm[32 * 256];
  __align__(16) unsigned char a_tr[128 * 64];
  __align__(16) unsigned char b_rm[32 * 256];
  __align__(16) unsigned char b_tr[128 * 64];
};

struct SmemFA {
  ImageSet img[2];                               // double buffer: 64 KB
  __align__(16) unsigned char pa[FA_WAVES][32 * 64];  // P transpose bufs
  __align__(16) unsigned char pb[FA_WAVES][32 * 64];  // dS transpose bufs
  // per-wave K/V operand tiles in A-fragment order (dkdv kernel only):
  // [wave][t16][lane][8 bf16] = one b128 per (t16, lane)
  __align__(16) unsigned char kv_ops[FA_WAVES][2][8 * 64 * 16];
};

struct TileRegs {
  uint4 a_lo, a_hi, b_lo, b_hi;
};

// issue the global loads for one (tileA, tileB) pair; thread t owns row
// q=t>>3, d-segment (t&7)*16..+16 of each 32x128 tile
__device__ inline TileRegs load_tiles(const bf16* __restrict__ srcA,
                                      const bf16* __restrict__ srcB) {
  const int t = threadIdx.x;
  const int64_t off = (int64_t)(t >> 3) * FA_D + (t & 7) * 16;
  TileRegs r;
  r.a_lo = reinterpret_cast<const uint4*>(srcA + off)[0];
  r.a_hi = reinterpret_cast<const uint4*>(srcA + off)[1];
  r.b_lo = reinterpret_cast<const uint4*>(srcB + off)[0];
  r.b_hi = reinterpret_cast<const uint4*>(srcB + off)[1];
  return r;
}

template <bool WRITE_TR = true>
__device__ inline void write_one(unsigned char* rm, unsigned char* tr, int q,
                                 int dseg, uint4 lo, uint4 hi) {
  *reinterpret_cast<uint4*>(rm + rm_addr(q, dseg * 2)) = lo;
  *reinterpret_cast<uint4*>(rm + rm_addr(q, dseg * 2 + 16)) = hi;
  if constexpr (!WRITE_TR) return;
  // b16 transpose scatter. (A paired-b32 variant using 2 shuffles per
  // element was tried and REGRESSED 17.6 -> 22.0 ms: the VALU shuffle cost
  // exceeds the saved LDS write issue. Next lever is ds_read_b64_tr_b16 on
  // the row-major image, which removes this image entirely.)
  const short* v1 = reinterpret_cast<const short*>(&lo);
  const short* v2 = reinterpret_cast<const short*>(&hi);
#pragma unroll
  for (int m = 0; m < 8; m++) {
    *reinterpret_cast<short*>(tr + tr_addr(dseg + m, q * 2)) = v1[m];
    *reinterpret_cast<short*>(tr + tr_addr(dseg + 8 + m, q * 2)) = v2[m];
  }
}

template <bool B_TR = true>
__device__ inline void write_tiles(ImageSet* img, const TileRegs& r) {
  const int t = threadIdx.x;
  const int q = t >> 3;
  const int dseg = (t & 7) * 16;
  write_one(img->a_rm, img->a_tr, q, dseg, r.a_lo, r.a_hi);
  // the dq kernel never reads the transposed V image — skip its scatter
  write_one<B_TR>(img->b_rm, img->b_tr, q, dseg, r.b_lo, r.b_hi);
}

__device__ inline bf16x8_vec rm_bfrag(const unsigned char* rm, int t16, int half,
                                      int l31) {
  return *reinterpret_cast<const bf16x8_vec*>(rm + rm_addr(l31, t16 * 32 + half * 16));
}
__device__ inline bf16x8_vec tr_bfrag(const unsigned char* tr, int dt, int h2,
                                      int half, int l31) {
  return *reinterpret_cast<const bf16x8_vec*>(
      tr + tr_addr(dt * 32 + l31, h2 * 32 + half * 16));
}
__device__ inline bf16x8_vec pb_afrag(const unsigned char* pb, int h2, int half,
                                      int l31) {
  return *reinterpret_cast<const bf16x8_vec*>(pb + pb_addr(l31, h2 * 32 + half * 16));
}
__device__ inline int c_row(int reg, int half) {
  return (reg & 3) + 8 * (reg >> 2) + 4 * half;
}

// ---- kernel 1: dK/dV ------------------------------------------------------
__global__ __launch_bounds__(FA_THREADS, 1) void fa_bwd_dkdv_kernel(
    const bf16* __restrict__ q, const bf16* __restrict__ k,
    const bf16* __restrict__ v, const bf16* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    bf16* __restrict__ dk, bf16* __restrict__ dv, int B, int Hq, int Hkv,
    int S, float scale, int causal) {
  __shared__ SmemFA sm;
  const int G = Hq / Hkv;
  const int b = blockIdx.y / Hkv;
  const int hkv = blockIdx.y % Hkv;
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int half = lane >> 5;
  const int l31 = lane & 31;
  const int jb = blockIdx.x * FA_WAVES + wave;
  const bool active = jb * FA_BLK < S;
  const int nQ = S / FA_BLK;
  const int i_min = causal ? blockIdx.x * FA_WAVES : 0;
  const int nI = nQ - i_min;
  const int T = G * nI;  // flattened (g, i) iteration count

  // K/V operand fragments live in wave-private LDS (A-fragment order, one
  // b128 per use) — holding them in VGPRs capped the allocator at 256 and
  // serialized every B-fragment ds_read behind an lgkmcnt(0)
  unsigned char* kops = sm.kv_ops[wave][0];
  unsigned char* vops = sm.kv_ops[wave][1];
  if (active) {
    const int64_t kv_off = (((int64_t)b * Hkv + hkv) * S + jb * FA_BLK + l31) * FA_D;
#pragma unroll
    for (int t = 0; t < 8; t++) {
      *reinterpret_cast<bf16x8_vec*>(kops + (t * 64 + lane) * 16) =
          *reinterpret_cast<const bf16x8_vec*>(k + kv_off + t * 16 + half * 8);
      *reinterpret_cast<bf16x8_vec*>(vops + (t * 64 + lane) * 16) =
          *reinterpret_cast<const bf16x8_vec*>(v + kv_off + t * 16 + half * 8);
    }
  }

  f32x16 dk_acc[4] = {};
  f32x16 dv_acc[4] = {};

  const int64_t head_stride = (int64_t)S * FA_D;
  const bf16* q_base = q + ((int64_t)b * Hq + hkv * G) * head_stride;
  const bf16* do_base = dout + ((int64_t)b * Hq + hkv * G) * head_stride;
  const float* lse_base = lse + ((int64_t)b * Hq + hkv * G) * S;
  const float* delta_base = delta + ((int64_t)b * Hq + hkv * G) * S;

  auto tile_src = [&](int t, const bf16* base) -> const bf16* {
    const int g = t / nI;
    const int i = i_min + t % nI;
    return base + (int64_t)g * head_stride + (int64_t)i * FA_BLK * FA_D;
  };

  for (int t = 0; t < T; t++) {
    const int cur = t & 1;
    const int g = t / nI;
    const int i = i_min + t % nI;

    // stage this tile (registers freed immediately — keeping a cross-
    // iteration register pipeline pushed the allocator to the VGPR cap and
    // serialized every B-fragment ds_read behind lgkmcnt(0))
    {
      TileRegs r = load_tiles(tile_src(t, q_base), tile_src(t, do_base));
      write_tiles(&sm.img[cur], r);
    }
    __syncthreads();

    if (active && !(causal && i < jb)) {
      const ImageSet* img = &sm.img[cur];
      const float lse_q = lse_base[(int64_t)g * S + i * FA_BLK + l31];
      const float delta_q = delta_base[(int64_t)g * S + i * FA_BLK + l31];

      // interleaved S^T / dP^T chains (independent accumulators)
      f32x16 s_acc = {};
      f32x16 dp_acc = {};
#pragma unroll
      for (int tt = 0; tt < 8; tt++) {
        const bf16x8_vec kf =
            *reinterpret_cast<const bf16x8_vec*>(kops + (tt * 64 + lane) * 16);
        const bf16x8_vec vf =
            *reinterpret_cast<const bf16x8_vec*>(vops + (tt * 64 + lane) * 16);
        s_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            kf, rm_bfrag(img->a_rm, tt, half, l31), s_acc, 0, 0, 0);
        dp_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            vf, rm_bfrag(img->b_rm, tt, half, l31), dp_acc, 0, 0, 0);
      }

      const int qg = i * FA_BLK + l31;
      unsigned char* pa = sm.pa[wave];
      unsigned char* pb = sm.pb[wave];
#pragma unroll
      for (int rg = 0; rg < 16; rg++) {
        const int row = c_row(rg, half);
        const int kg = jb * FA_BLK + row;
        const bool valid = !causal || (qg >= kg);
        const float pv = valid ? __expf(scale * s_acc[rg] - lse_q) : 0.0f;
        *reinterpret_cast<bf16*>(pa + pb_addr(row, l31 * 2)) = __float2bfloat16(pv);
        *reinterpret_cast<bf16*>(pb + pb_addr(row, l31 * 2)) =
            __float2bfloat16(pv * (dp_acc[rg] - delta_q) * scale);
      }

      // dV += P^T dO ; dK += dS^T Q — 16 MFMAs over 8 independent accs
#pragma unroll
      for (int dt = 0; dt < 4; dt++) {
#pragma unroll
        for (int h2 = 0; h2 < 2; h2++) {
          dv_acc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              pb_afrag(pa, h2, half, l31), tr_bfrag(img->b_tr, dt, h2, half, l31),
              dv_acc[dt], 0, 0, 0);
          dk_acc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              pb_afrag(pb, h2, half, l31), tr_bfrag(img->a_tr, dt, h2, half, l31),
              dk_acc[dt], 0, 0, 0);
        }
      }
    }
    __syncthreads();
  }

  if (!active) return;
  const int64_t kv_off = (((int64_t)b * Hkv + hkv) * S + jb * FA_BLK) * FA_D;
#pragma unroll
  for (int dt = 0; dt < 4; dt++) {
#pragma unroll
    for (int rg = 0; rg < 16; rg++) {
      const int key = c_row(rg, half);
      const int d = dt * 32 + l31;
      dk[kv_off + (int64_t)key * FA_D + d] = __float2bfloat16(dk_acc[dt][rg]);
      dv[kv_off + (int64_t)key * FA_D + d] = __float2bfloat16(dv_acc[dt][rg]);
    }
  }
}

// ---- kernel 2: dQ ---------------------------------------------------------
__global__ __launch_bounds__(FA_THREADS, 1) void fa_bwd_dq_kernel(
    const bf16* __restrict__ q, const bf16* __restrict__ k,
    const bf16* __restrict__ v, const bf16* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    bf16* __restrict__ dq, int B, int Hq, int Hkv, int S, float scale,
    int causal) {
  __shared__ SmemFA sm;
  const int G = Hq / Hkv;
  const int b = blockIdx.y / Hq;
  const int hq = blockIdx.y % Hq;
  const int hkv = hq / G;
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int half = lane >> 5;
  const int l31 = lane & 31;
  const int ib = blockIdx.x * FA_WAVES + wave;
  const bool active = ib * FA_BLK < S;
  const int nK = S / FA_BLK;
  const int T = causal ? min(blockIdx.x * FA_WAVES + FA_WAVES, nK) : nK;

  bf16x8_vec qfrag[8], dofrag[8];
  float lse_row[16], delta_row[16];
  if (active) {
    const int64_t q_off = (((int64_t)b * Hq + hq) * S + ib * FA_BLK + l31) * FA_D;
#pragma unroll
    for (int t = 0; t < 8; t++) {
      qfrag[t] = *reinterpret_cast<const bf16x8_vec*>(q + q_off + t * 16 + half * 8);
      dofrag[t] = *reinterpret_cast<const bf16x8_vec*>(dout + q_off + t * 16 + half * 8);
    }
    const float* lse_head = lse + ((int64_t)b * Hq + hq) * S;
    const float* delta_head = delta + ((int64_t)b * Hq + hq) * S;
#pragma unroll
    for (int rg = 0; rg < 16; rg++) {
      const int row = ib * FA_BLK + c_row(rg, half);
      lse_row[rg] = lse_head[row];
      delta_row[rg] = delta_head[row];
    }
  }

  f32x16 dq_acc[4] = {};

  const bf16* k_head = k + ((int64_t)b * Hkv + hkv) * S * FA_D;
  const bf16* v_head = v + ((int64_t)b * Hkv + hkv) * S * FA_D;

  for (int j = 0; j < T; j++) {
    const int cur = j & 1;
    {
      TileRegs r = load_tiles(k_head + (int64_t)j * FA_BLK * FA_D,
                              v_head + (int64_t)j * FA_BLK * FA_D);
      write_tiles<false>(&sm.img[cur], r);
    }
    __syncthreads();

    if (active && !(causal && j > ib)) {
      const ImageSet* img = &sm.img[cur];
      f32x16 s_acc = {};
      f32x16 dp_acc = {};
#pragma unroll
      for (int tt = 0; tt < 8; tt++) {
        s_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            qfrag[tt], rm_bfrag(img->a_rm, tt, half, l31), s_acc, 0, 0, 0);
        dp_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            dofrag[tt], rm_bfrag(img->b_rm, tt, half, l31), dp_acc, 0, 0, 0);
      }

      unsigned char* pb = sm.pb[wave];
      const int kg = j * FA_BLK + l31;
#pragma unroll
      for (int rg = 0; rg < 16; rg++) {
        const int qg = ib * FA_BLK + c_row(rg, half);
        const bool valid = !causal || (qg >= kg);
        const float p = valid ? __expf(scale * s_acc[rg] - lse_row[rg]) : 0.0f;
        const float ds = p * (dp_acc[rg] - delta_row[rg]) * scale;
        *reinterpret_cast<bf16*>(pb + pb_addr(c_row(rg, half), l31 * 2)) =
            __float2bfloat16(ds);
      }

#pragma unroll
      for (int dt = 0; dt < 4; dt++) {
#pragma unroll
        for (int h2 = 0; h2 < 2; h2++) {
          dq_acc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              pb_afrag(pb, h2, half, l31), tr_bfrag(img->a_tr, dt, h2, half, l31),
              dq_acc[dt], 0, 0, 0);
        }
      }
    }
    __syncthreads();
  }

  if (!active) return;
  const int64_t q_off = (((int64_t)b * Hq + hq) * S + ib * FA_BLK) * FA_D;
#pragma unroll
  for (int dt = 0; dt < 4; dt++) {
#pragma unroll
    for (int rg = 0; rg < 16; rg++) {
      dq[q_off + (int64_t)c_row(rg, half) * FA_D + dt * 32 + l31] =
          __float2bfloat16(dq_acc[dt][rg]);
    }
  }
}

// ---- launchers ------------------------------------------------------------

void launch_fa_bwd(const void* q, const void* k, const void* v, const void* dout,
                   const float* lse, const float* delta, void* dq, void* dk,
                   void* dv, int B, int Hq, int Hkv, int S, float scale,
                   bool causal, hipStream_t stream) {
  const int nblk = (S + FA_BLK * FA_WAVES - 1) / (FA_BLK * FA_WAVES);
  hipLaunchKernelGGL(fa_bwd_dkdv_kernel, dim3(nblk, B * Hkv), dim3(FA_THREADS), 0,
                     stream, (const bf16*)q, (const bf16*)k, (const bf16*)v,
                     (const bf16*)dout, lse, delta, (bf16*)dk, (bf16*)dv, B, Hq,
                     Hkv, S, scale, causal ? 1 : 0);
  hipLaunchKernelGGL(fa_bwd_dq_kernel, dim3(nblk, B * Hq), dim3(FA_THREADS), 0,
                     stream, (const bf16*)q, (const bf16*)k, (const bf16*)v,
                     (const bf16*)dout, lse, delta, (bf16*)dq, B, Hq, Hkv, S,
                     scale, causal ? 1 : 0);
}

}  // namespace torchft_amd
