// Flash-attention FORWARD for CDNA4 (gfx950) — bf16, D=128, causal+full,
// GQA-native. Produces O and the logsumexp the in-tree backward consumes
// (P = exp(scale*S - LSE)), replacing aten/aotriton's attn_fwd.
//
// Structure (guide §B "fused attention prefill", adapted):
//   grid (S/256, B*Hq); one 512-thread workgroup = 8 waves, wave w owns
//   q rows [Q0 + 32w, +32) of one query head. K/V stream through
//   double-buffered LDS tiles of KVBLK=64 keys, staged cooperatively and
//   SHARED by all 8 waves (same kv head under GQA).
//
//   Swapped QK^T: S_blk = mfma(A=K, B=Q^T) puts q in the MFMA column
//   (lane l31 = its q row), so the whole online softmax — running max m,
//   denominator l, P = exp(scale*s - m) — is lane-local per q: the lane
//   pair (l, l+32) holds the 2x16 key rows of one q and combines via
//   permlane32_swap. P converts to the PV A-fragment in-register with
//   v_cvt_pk_bf16_f32 + permlane32_swap (no LDS round trip, T12).
//
//   PV: B-fragments of V by ds_read_b64_tr_b16 hardware-transpose reads
//   from the row-major subtiled V image (the bwd kernel's recipe; probe:
//   tr16_probe.hip). O accumulates [16 q-rows][d=l31] per 32-d tile.
//
//   Online-softmax rescale: O's rows are q = c_row(reg) while m/l live at
//   lane q = l31, so a rescale needs an alpha broadcast through LDS. With
//   the defer-max threshold (T13, THR=8 in scaled-score units) the branch
//   is rare: the common path multiplies nothing. At a rescale EVERYTHING
//   still at the old max is scaled exactly once: the decision happens
//   before this block's P is exponentiated, and l folds alpha in the same
//   update (l = l*alpha + sum_p).
//
// MFMA lane mappings (mfma_probe.hip):
//   A: row=lane&31, k=(lane>>5)*8+m   B: k=(lane>>5)*8+m, col=lane&31
//   C/D: col=lane&31, row=(reg&3)+8*(reg>>2)+4*(lane>>5)

#include <hip/hip_bf16.h>
#include <hip/hip_runtime.h>

#include <cstdint>

namespace torchft_amd {

using bf16 = __hip_bfloat16;
typedef __attribute__((__vector_size__(8 * sizeof(short)))) short bf16x8_vec;
typedef __attribute__((__vector_size__(16 * sizeof(float)))) float f32x16;

#define FWD_D 128
#define FWD_QBLK 32
#define FWD_KVBLK 64
#define FWD_WAVES 8
#define FWD_THREADS 512
// V/K subtile: [64 rows][16 cols] bf16 + 16-B pad -> stride 2080:
// 2080/4 = 520 ≡ 8 (mod 32): the 8 staging ds_write_b128 of a lane group
// hit 4 distinct bank quads (2-way); ≡ 8 (mod 64) keeps adjacent-subtile
// tr blocks mostly disjoint.
#define FSUBT 2080
#define FWD_THR 11.54f  // defer-max threshold: 8 nats in log2 units
#define LOG2E 1.44269504089f

struct FwdTile {
  __align__(16) unsigned char sub[8 * FSUBT];
};

struct SmemFwd {
  FwdTile k_img[2];
  FwdTile v_img[2];
  __align__(16) float bcast[FWD_WAVES][32];  // alpha / 1/l broadcasts
};

__device__ inline int fst_addr(int tt, int row, int byte_in_row) {
  return tt * FSUBT + row * 32 + byte_in_row;
}

// T14 async-STAGE split: thread t of 512 owns rows r=t>>3, 32-B chunk
// pair c0=(t&7)*2 of each [64][128] bf16 tile. The loads issue an
// iteration EARLY (under the previous tile's MFMA phase, hiding the HBM
// latency); the writes land after the barrier that frees the buffer.
struct KvRegs {
  uint4 k0, k1, v0, v1;
};

__device__ inline KvRegs load_kv(const bf16* __restrict__ ksrc,
                                 const bf16* __restrict__ vsrc, int64_t ld) {
  const int t = threadIdx.x;
  const int64_t off = (int64_t)(t >> 3) * ld + ((t & 7) * 2) * 8;
  KvRegs r;
  r.k0 = reinterpret_cast<const uint4*>(ksrc + off)[0];
  r.k1 = reinterpret_cast<const uint4*>(ksrc + off)[1];
  r.v0 = reinterpret_cast<const uint4*>(vsrc + off)[0];
  r.v1 = reinterpret_cast<const uint4*>(vsrc + off)[1];
  return r;
}

__device__ inline void write_kv(FwdTile* kd, FwdTile* vd, const KvRegs& r) {
  const int t = threadIdx.x;
  const int c0 = (t & 7) * 2;
  const int a0 = fst_addr(c0 >> 1, t >> 3, (c0 & 1) * 16);
  *reinterpret_cast<uint4*>(kd->sub + a0) = r.k0;
  *reinterpret_cast<uint4*>(kd->sub + a0 + 16) = r.k1;
  *reinterpret_cast<uint4*>(vd->sub + a0) = r.v0;
  *reinterpret_cast<uint4*>(vd->sub + a0 + 16) = r.v1;
}

// K A-fragment: row = key (s*32 + l31), k-dim = d slice tt: contiguous b128
__device__ inline bf16x8_vec k_afrag(const unsigned char* img, int tt, int s,
                                     int half, int l31) {
  return *reinterpret_cast<const bf16x8_vec*>(
      img + fst_addr(tt, s * 32 + l31, half * 16));
}

__device__ inline int c_row(int reg, int half) {
  return (reg & 3) + 8 * (reg >> 2) + 4 * half;
}

// tr_b16 per-lane invariant for V B-fragments (rows = keys in [64][16]
// subtiles): sub parity (l>>4)&1, key offset (l>>5)*8 rows, chunk (l&15)*8
__device__ inline unsigned fwd_tr_lane_off(int lane) {
  return ((lane >> 4) & 1) * FSUBT + ((lane >> 5) * 8) * 32 + (lane & 15) * 8;
}

#define FTR_READ(dst, addr) \
  asm volatile("ds_read_b64_tr_b16 %0, %1" : "=v"(dst) : "v"(addr))
#define FTR_WAIT2(r0, r1) \
  asm volatile("s_waitcnt lgkmcnt(0)" : "+v"(r0), "+v"(r1))

// sequence-row offset of head (b,h): bshd=1 = [B,S,H,D] storage (see
// flash_attn_bwd.hip row_off)
__device__ inline int64_t frow_off(int b, int h, int s, int H, int S, int bshd) {
  return bshd ? (((int64_t)b * S + s) * H + h) * FWD_D
              : (((int64_t)b * H + h) * S + s) * FWD_D;
}

__device__ inline bf16x8_vec tr_join8(unsigned long long lo,
                                      unsigned long long hi) {
  union {
    struct {
      unsigned long long lo, hi;
    } u;
    bf16x8_vec v;
  } c;
  c.u.lo = lo;
  c.u.hi = hi;
  return c.v;
}

// pack two f32 into one dword of 2 bf16 (no builtin on gfx950 — T12).
// The trailing s_nop provides the 2 wait states a later v_permlane32_swap
// needs after a VALU write of its operand — the compiler's hazard
// recognizer cannot see inside this asm (T21 hazard note).
__device__ inline unsigned cvt_pk_bf16(float lo, float hi) {
  unsigned r;
  asm("v_cvt_pk_bf16_f32 %0, %1, %2\n\ts_nop 1" : "=v"(r) : "v"(lo), "v"(hi));
  return r;
}

// exchange the 32-lane halves of two dwords (see T21):
// r0 = [a(0:31) | b(0:31)], r1 = [a(32:63) | b(32:63)]
__device__ inline void half_swap(unsigned& a, unsigned& b) {
  auto r = __builtin_amdgcn_permlane32_swap(a, b, false, false);
  a = r[0];
  b = r[1];
}

__device__ inline float half_combine_max(float v) {
  unsigned a = __float_as_uint(v), b = a;
  half_swap(a, b);
  return fmaxf(__uint_as_float(a), __uint_as_float(b));
}

__device__ inline float half_combine_sum(float v) {
  unsigned a = __float_as_uint(v), b = a;
  half_swap(a, b);
  return __uint_as_float(a) + __uint_as_float(b);
}

__global__ __launch_bounds__(FWD_THREADS, 1) void fa_fwd_kernel(
    const bf16* __restrict__ q, const bf16* __restrict__ k,
    const bf16* __restrict__ v, bf16* __restrict__ out,
    float* __restrict__ lse, int B, int Hq, int Hkv, int S, float scale,
    int causal, int bshd) {
  __shared__ SmemFwd sm;
  const int G = Hq / Hkv;
  const int b = blockIdx.y / Hq;
  const int hq = blockIdx.y % Hq;
  const int hkv = hq / G;
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int half = lane >> 5;
  const int l31 = lane & 31;

  const int Q0 = blockIdx.x * (FWD_WAVES * FWD_QBLK);
  const int my_q0 = Q0 + wave * FWD_QBLK;       // this wave's 32 q rows
  const int q_glob = my_q0 + l31;               // this lane's q row
  const bool active = my_q0 < S;

  // Q block in registers: lane holds Q[q_glob][tt*16 + half*8 .. +8]
  const int64_t lse_head = ((int64_t)b * Hq + hq) * S;  // lse is dense BHS
  bf16x8_vec qfrag[8];
  if (active) {
    const int64_t q_off = frow_off(b, hq, q_glob, Hq, S, bshd);
#pragma unroll
    for (int tt = 0; tt < 8; tt++) {
      qfrag[tt] =
          *reinterpret_cast<const bf16x8_vec*>(q + q_off + tt * 16 + half * 8);
    }
  }

  const bf16* k_base = k + frow_off(b, hkv, 0, Hkv, S, bshd);
  const bf16* v_base = v + frow_off(b, hkv, 0, Hkv, S, bshd);
  const int64_t kv_ld = bshd ? (int64_t)Hkv * FWD_D : FWD_D;

  // number of KV tiles this WORKGROUP must stage
  const int kv_hi = causal ? min(S, Q0 + FWD_WAVES * FWD_QBLK) : S;
  const int nT = (kv_hi + FWD_KVBLK - 1) / FWD_KVBLK;

  f32x16 o_acc[4] = {};
  // softmax runs in the log2 domain: m2 = max of (scale*log2e)*s, powers
  // of two via the native v_exp, one fma per element (exp2+fma fold)
  const float c2 = scale * LOG2E;
  float m_run = -1e30f;
  float l_run = 0.f;

  // the second-dispatched wave half loses VALU arbitration on every
  // segment; one static priority raise for it (T5 static form — the
  // condition must be provably wave-uniform or s_setprio goes under exec)
  if (__builtin_amdgcn_readfirstlane(threadIdx.x) >= 256) {
    __builtin_amdgcn_s_setprio(1);
  }

  const unsigned tr_off = fwd_tr_lane_off(lane);
  float* bc = sm.bcast[wave];

  KvRegs staged = load_kv(k_base, v_base, kv_ld);  // tile 0
  for (int j = 0; j < nT; j++) {
    const int cur = j & 1;
    write_kv(&sm.k_img[cur], &sm.v_img[cur], staged);
    __syncthreads();
    if (j + 1 < nT) {
      // issue the next tile's global loads now — they retire under this
      // tile's MFMA phase and are waited for by the write after the
      // next barrier (compiler-counted vmcnt)
      staged = load_kv(k_base + (int64_t)(j + 1) * FWD_KVBLK * kv_ld,
                       v_base + (int64_t)(j + 1) * FWD_KVBLK * kv_ld, kv_ld);
    }

    const int key0 = j * FWD_KVBLK;
    // this wave needs the tile only if some of its keys are visible
    if (active && (!causal || key0 <= my_q0 + FWD_QBLK - 1)) {
      const unsigned char* kimg = sm.k_img[cur].sub;
      const unsigned v_img_base = (unsigned)(uintptr_t)sm.v_img[cur].sub + tr_off;

#pragma unroll
      for (int s = 0; s < 2; s++) {  // two 32-key blocks per tile
        const int kb0 = key0 + s * 32;
        if (causal && kb0 > my_q0 + FWD_QBLK - 1) break;

        f32x16 s_acc = {};
#pragma unroll
        for (int tt = 0; tt < 8; tt++) {
          s_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              k_afrag(kimg, tt, s, half, l31), qfrag[tt], s_acc, 0, 0, 0);
        }

        // raw scores + causal mask; per-lane block max over its q.
        // One multiply converts the max to the log2 domain (c2 > 0
        // preserves order); each element pays a single fma inside exp2.
        const bool diag = causal && (kb0 + 31 > q_glob);
        float p[16];
        float bmax = -1e30f;
#pragma unroll
        for (int r = 0; r < 16; r++) {
          float sv = s_acc[r];
          if (diag && (kb0 + c_row(r, half) > q_glob)) sv = -1e30f;
          p[r] = sv;
          bmax = fmaxf(bmax, sv);
        }
        bmax = half_combine_max(bmax) * c2;  // both 16-key halves of this q

        // defer-max: rescale O only when the max moved past THR
        float m_use = m_run;
        if (!__all(bmax - m_run <= FWD_THR)) {
          const float m_new = fmaxf(m_run, bmax);
          const float alpha = exp2f(m_run - m_new);  // 0 on first tile
          // broadcast alpha to O's row layout and rescale
          if (half == 0) bc[l31] = alpha;
          __builtin_amdgcn_wave_barrier();
          asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
#pragma unroll
          for (int dt = 0; dt < 4; dt++) {
#pragma unroll
            for (int r = 0; r < 16; r++) {
              o_acc[dt][r] *= bc[c_row(r, half)];
            }
          }
          l_run *= alpha;
          m_run = m_new;
          m_use = m_new;
        }

        // P = exp(s - m) (bounded by e^THR on the defer path), row sum,
        // and bf16 A-fragments for PV via cvt_pk + half swaps
        float psum = 0.f;
        unsigned pk[8];
#pragma unroll
        for (int i = 0; i < 8; i++) {
          const float e0 = exp2f(fmaf(p[2 * i], c2, -m_use));
          const float e1 = exp2f(fmaf(p[2 * i + 1], c2, -m_use));
          psum += e0 + e1;
          pk[i] = cvt_pk_bf16(e0, e1);
        }
        l_run += half_combine_sum(psum);

        // assemble A-fragments: lane needs P[q][keys (l>>5)*8 + 0..7] per
        // 16-key step h2; own regs hold keys c_row(r,half) — pair halves
        // swap so [pk0,pk2',pk1,pk3'] become consecutive key pairs
        half_swap(pk[0], pk[2]);  // keys {0,1}|{8,9} <-> {4,5}|{12,13}
        half_swap(pk[1], pk[3]);  // keys {2,3}|{10,11} <-> {6,7}|{14,15}
        half_swap(pk[4], pk[6]);  // second 16-key step
        half_swap(pk[5], pk[7]);
        union {
          unsigned u[4];
          bf16x8_vec v;
        } pa0, pa1;
        pa0.u[0] = pk[0];
        pa0.u[1] = pk[1];
        pa0.u[2] = pk[2];
        pa0.u[3] = pk[3];
        pa1.u[0] = pk[4];
        pa1.u[1] = pk[5];
        pa1.u[2] = pk[6];
        pa1.u[3] = pk[7];

        // PV: V B-fragments by hardware transpose; k-dim = 16 keys per
        // h2. dt fastest so consecutive MFMAs rotate the four O
        // accumulators (dependent-latency hiding; see bwd kernels).
#pragma unroll
        for (int it = 0; it < 8; it++) {
          const int dt = it & 3;
          const int h2 = it >> 2;
          const unsigned base =
              v_img_base + dt * (2 * FSUBT) + (s * 32 + h2 * 16) * 32;
          unsigned long long t0, t1;
          FTR_READ(t0, base);
          FTR_READ(t1, base + 128);
          FTR_WAIT2(t0, t1);
          o_acc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              h2 ? pa1.v : pa0.v, tr_join8(t0, t1), o_acc[dt], 0, 0, 0);
        }
      }
    }
    __syncthreads();
  }

  if (!active) return;

  // epilogue: O /= l (l lives at lane q=l31; O rows are c_row) and store.
  // bc is wave-private and the write/read are same-wave DS ops, ordered by
  // the LDS pipeline — no barrier (and a workgroup barrier would deadlock:
  // inactive waves returned above).
  if (half == 0) bc[l31] = 1.f / l_run;
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
#pragma unroll
  for (int dt = 0; dt < 4; dt++) {
#pragma unroll
    for (int r = 0; r < 16; r++) {
      const int row = c_row(r, half);
      out[frow_off(b, hq, my_q0 + row, Hq, S, bshd) + dt * 32 + l31] =
          __float2bfloat16(o_acc[dt][r] * bc[row]);
    }
  }
  if (half == 0) {
    // convert the log2-domain state back to the natural-log lse the
    // backward consumes: lse = ln2 * (m2 + log2(l))
    lse[lse_head + q_glob] = 0.6931471805599453f * (m_run + __log2f(l_run));
  }
}

void launch_fa_fwd(const void* q, const void* k, const void* v, void* out,
                   float* lse, int B, int Hq, int Hkv, int S, float scale,
                   bool causal, bool bshd, hipStream_t stream) {
  const int rows_per_wg = FWD_WAVES * FWD_QBLK;
  const int nblk = (S + rows_per_wg - 1) / rows_per_wg;
  hipLaunchKernelGGL(fa_fwd_kernel, dim3(nblk, B * Hq), dim3(FWD_THREADS), 0,
                     stream, (const bf16*)q, (const bf16*)k, (const bf16*)v,
                     (bf16*)out, lse, B, Hq, Hkv, S, scale, causal ? 1 : 0,
                     bshd ? 1 : 0);
}

}  // namespace torchft_amd
