// torch-extension bindings for the CDNA4 kernels (fp8 quantized-collective
// trio, fused model ops, fused AdamW). Tensor-level API consumed by
// torchft_amd.ops / torchft_amd.quantization.
#include <ATen/cuda/CUDAContext.h>
#include <torch/extension.h>

#include <cmath>

#include <vector>

#include "kernels.h"

namespace tft = torchft_amd;

namespace {

constexpr int64_t kQBlock = 2048;

struct PackGeom {
  int64_t total_blocks = 0;
  int64_t padded_blocks = 0;
  int64_t blocks_per_rank = 0;
  int64_t slice_bytes = 0;
};

PackGeom pack_geometry(const std::vector<at::Tensor>& tensors, int64_t world) {
  PackGeom g;
  for (auto& t : tensors) g.total_blocks += (t.numel() + kQBlock - 1) / kQBlock;
  g.padded_blocks = ((g.total_blocks + world - 1) / world) * world;
  g.blocks_per_rank = g.padded_blocks / world;
  g.slice_bytes = g.blocks_per_rank * (4 + kQBlock);
  return g;
}

// Builds the device-side metadata arrays (ptrs, block prefix, numels).
struct DeviceMeta {
  at::Tensor ptrs, prefix, numels;
};

DeviceMeta build_meta(const std::vector<at::Tensor>& tensors) {
  const int64_t n = (int64_t)tensors.size();
  auto opts = at::TensorOptions().dtype(at::kLong).pinned_memory(true);
  at::Tensor host = at::empty({3 * n + 1}, opts);
  int64_t* h = host.data_ptr<int64_t>();
  int64_t* ptrs = h;
  int64_t* numels = h + n;
  int64_t* prefix = h + 2 * n;  // n+1 entries
  int64_t acc = 0;
  for (int64_t i = 0; i < n; i++) {
    auto& t = tensors[i];
    TORCH_CHECK(t.is_contiguous(), "fp8 pack requires contiguous tensors");
    TORCH_CHECK(t.is_cuda(), "fp8 pack requires device tensors");
    ptrs[i] = (int64_t)t.data_ptr();
    numels[i] = t.numel();
    prefix[i] = acc;
    acc += (t.numel() + kQBlock - 1) / kQBlock;
  }
  prefix[n] = acc;
  at::Tensor dev = host.to(tensors[0].device(), /*non_blocking=*/true);
  DeviceMeta m;
  m.ptrs = dev.narrow(0, 0, n);
  m.numels = dev.narrow(0, n, n);
  m.prefix = dev.narrow(0, 2 * n, n + 1);
  return m;
}

}  // namespace

// ---- fp8 quantized-collective support --------------------------------------

int64_t fp8_pack_bytes(const std::vector<at::Tensor>& tensors, int64_t world) {
  auto g = pack_geometry(tensors, world);
  return g.slice_bytes * world;
}

int64_t fp8_slice_bytes(const std::vector<at::Tensor>& tensors, int64_t world) {
  return pack_geometry(tensors, world).slice_bytes;
}

void fp8_quantize(const std::vector<at::Tensor>& tensors, at::Tensor pack,
                  int64_t world) {
  TORCH_CHECK(!tensors.empty(), "need at least one tensor");
  auto g = pack_geometry(tensors, world);
  TORCH_CHECK(pack.numel() >= g.slice_bytes * world, "pack buffer too small");
  auto meta = build_meta(tensors);
  auto stream = at::cuda::getCurrentCUDAStream().stream();
  auto st = tensors[0].scalar_type();
  for (auto& t : tensors) TORCH_CHECK(t.scalar_type() == st, "mixed dtypes");
  const int64_t* ptrs = meta.ptrs.data_ptr<int64_t>();
  const int64_t* prefix = meta.prefix.data_ptr<int64_t>();
  const int64_t* numels = meta.numels.data_ptr<int64_t>();
  uint8_t* p = pack.data_ptr<uint8_t>();
  int code = st == at::kBFloat16 ? 0 : st == at::kHalf ? 1 : st == at::kFloat ? 2 : -1;
  TORCH_CHECK(code >= 0, "unsupported dtype for fp8 quantize: ", st);
  tft::launch_quantize_dtype(code, ptrs, prefix, numels, (int)tensors.size(),
                             g.total_blocks, g.padded_blocks, g.blocks_per_rank,
                             g.slice_bytes, p, (tft_stream)stream);
}

void fp8_dequantize(const std::vector<at::Tensor>& tensors, at::Tensor pack,
                    int64_t world) {
  TORCH_CHECK(!tensors.empty(), "need at least one tensor");
  auto g = pack_geometry(tensors, world);
  auto meta = build_meta(tensors);
  auto stream = at::cuda::getCurrentCUDAStream().stream();
  auto st = tensors[0].scalar_type();
  const int64_t* ptrs = meta.ptrs.data_ptr<int64_t>();
  const int64_t* prefix = meta.prefix.data_ptr<int64_t>();
  const int64_t* numels = meta.numels.data_ptr<int64_t>();
  const uint8_t* p = pack.data_ptr<uint8_t>();
  int code = st == at::kBFloat16 ? 0 : st == at::kHalf ? 1 : st == at::kFloat ? 2 : -1;
  TORCH_CHECK(code >= 0, "unsupported dtype for fp8 dequantize: ", st);
  tft::launch_dequantize_dtype(code, ptrs, prefix, numels, (int)tensors.size(),
                               g.total_blocks, g.padded_blocks, g.blocks_per_rank,
                               g.slice_bytes, p, (tft_stream)stream);
}

void fp8_reduce(at::Tensor recv, at::Tensor out, int64_t world, bool avg) {
  TORCH_CHECK(recv.is_cuda() && out.is_cuda());
  TORCH_CHECK(recv.numel() == out.numel() * world, "recv must hold world slices");
  const int64_t slice_bytes = out.numel();
  const int64_t blocks_per_rank = slice_bytes / (4 + kQBlock);
  TORCH_CHECK(blocks_per_rank * (4 + kQBlock) == slice_bytes, "bad slice size");
  auto stream = at::cuda::getCurrentCUDAStream().stream();
  tft::launch_reduce(recv.data_ptr<uint8_t>(), out.data_ptr<uint8_t>(), (int)world,
                     blocks_per_rank, slice_bytes, avg, (tft_stream)stream);
}

// ---- fused model ops --------------------------------------------------------

std::vector<at::Tensor> rmsnorm_fwd(at::Tensor x, at::Tensor w, double eps) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16 && x.is_contiguous());
  TORCH_CHECK(w.is_cuda() && w.scalar_type() == at::kBFloat16 && w.is_contiguous());
  const int H = (int)x.size(-1);
  TORCH_CHECK(H % 8 == 0, "hidden dim must be a multiple of 8");
  const int64_t rows = x.numel() / H;
  auto y = at::empty_like(x);
  auto invrms = at::empty({rows}, x.options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentCUDAStream().stream();
  tft::launch_rmsnorm_fwd(x.data_ptr(), w.data_ptr(), y.data_ptr(),
                          invrms.data_ptr<float>(), rows, H, (float)eps,
                          (tft_stream)stream);
  return {y, invrms};
}

std::vector<at::Tensor> rmsnorm_bwd(at::Tensor dy, at::Tensor x, at::Tensor w,
                                    at::Tensor invrms) {
  const int H = (int)x.size(-1);
  const int64_t rows = x.numel() / H;
  TORCH_CHECK(H * sizeof(float) <= 160 * 1024, "H too large for LDS dw buffer");
  auto dx = at::empty_like(x);
  auto dw = at::zeros({H}, x.options().dtype(at::kFloat));
  auto dyc = dy.contiguous();
  auto stream = at::cuda::getCurrentCUDAStream().stream();
  tft::launch_rmsnorm_bwd(dyc.data_ptr(), x.data_ptr(), w.data_ptr(),
                          invrms.data_ptr<float>(), dx.data_ptr(),
                          dw.data_ptr<float>(), rows, H, (tft_stream)stream);
  return {dx, dw};
}

at::Tensor rope_apply(at::Tensor x, at::Tensor cos_tab, at::Tensor sin_tab,
                      bool backward) {
  // x: [B, S, Hh, D]
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16 && x.dim() == 4);
  auto xc = x.contiguous();
  const int S = (int)x.size(1);
  const int Hh = (int)x.size(2);
  const int D = (int)x.size(3);
  TORCH_CHECK(D % 8 == 0, "head dim must be a multiple of 8");
  TORCH_CHECK(cos_tab.size(0) >= S && cos_tab.size(1) == D / 2, "cos table shape");
  auto out = at::empty_like(xc);
  const int64_t total_pairs = xc.numel() / 2;
  auto stream = at::cuda::getCurrentCUDAStream().stream();
  tft::launch_rope(xc.data_ptr(), out.data_ptr(), cos_tab.data_ptr<float>(),
                   sin_tab.data_ptr<float>(), total_pairs, S, Hh, D, backward,
                   (tft_stream)stream);
  return out;
}

at::Tensor swiglu_fwd(at::Tensor a, at::Tensor b) {
  TORCH_CHECK(a.is_cuda() && a.scalar_type() == at::kBFloat16);
  TORCH_CHECK(a.numel() % 8 == 0, "numel must be a multiple of 8");
  auto ac = a.contiguous();
  auto bc = b.contiguous();
  auto out = at::empty_like(ac);
  auto stream = at::cuda::getCurrentCUDAStream().stream();
  tft::launch_swiglu_fwd(ac.data_ptr(), bc.data_ptr(), out.data_ptr(), ac.numel(),
                         (tft_stream)stream);
  return out;
}

at::Tensor swiglu_glu_fwd(at::Tensor gu) {
  TORCH_CHECK(gu.is_cuda() && gu.scalar_type() == at::kBFloat16);
  auto guc = gu.contiguous();
  const int64_t f = guc.size(-1) / 2;
  const int64_t rows = guc.numel() / (2 * f);
  TORCH_CHECK(f % 8 == 0, "glu width must be a multiple of 8");
  auto sizes = guc.sizes().vec();
  sizes.back() = f;
  auto out = at::empty(sizes, guc.options());
  auto stream = at::cuda::getCurrentCUDAStream().stream();
  tft::launch_swiglu_glu_fwd(guc.data_ptr(), out.data_ptr(), rows, f,
                             (tft_stream)stream);
  return out;
}

at::Tensor swiglu_glu_bwd(at::Tensor dy, at::Tensor gu) {
  auto dyc = dy.contiguous();
  auto guc = gu.contiguous();
  const int64_t f = guc.size(-1) / 2;
  const int64_t rows = guc.numel() / (2 * f);
  auto dgu = at::empty_like(guc);
  auto stream = at::cuda::getCurrentCUDAStream().stream();
  tft::launch_swiglu_glu_bwd(dyc.data_ptr(), guc.data_ptr(), dgu.data_ptr(),
                             rows, f, (tft_stream)stream);
  return dgu;
}

std::vector<at::Tensor> swiglu_bwd(at::Tensor dy, at::Tensor a, at::Tensor b) {
  auto dyc = dy.contiguous();
  auto ac = a.contiguous();
  auto bc = b.contiguous();
  auto da = at::empty_like(ac);
  auto db = at::empty_like(bc);
  auto stream = at::cuda::getCurrentCUDAStream().stream();
  tft::launch_swiglu_bwd(dyc.data_ptr(), ac.data_ptr(), bc.data_ptr(),
                         da.data_ptr(), db.data_ptr(), ac.numel(),
                         (tft_stream)stream);
  return {da, db};
}

// ---- fused AdamW ------------------------------------------------------------

void adamw_step(const std::vector<at::Tensor>& params,
                const std::vector<at::Tensor>& grads,
                const std::vector<at::Tensor>& exp_avgs,
                const std::vector<at::Tensor>& exp_avg_sqs, double lr,
                double beta1, double beta2, double eps, double weight_decay,
                int64_t step) {
  const int64_t n = (int64_t)params.size();
  TORCH_CHECK(n > 0);
  auto opts = at::TensorOptions().dtype(at::kLong).pinned_memory(true);
  at::Tensor host = at::empty({6 * n + 1}, opts);
  int64_t* h = host.data_ptr<int64_t>();
  int64_t acc = 0;
  for (int64_t i = 0; i < n; i++) {
    TORCH_CHECK(params[i].is_contiguous() && grads[i].is_contiguous());
    TORCH_CHECK(params[i].scalar_type() == at::kBFloat16, "params must be bf16");
    TORCH_CHECK(exp_avgs[i].scalar_type() == at::kFloat, "states must be fp32");
    h[i] = (int64_t)params[i].data_ptr();
    h[n + i] = (int64_t)grads[i].data_ptr();
    h[2 * n + i] = (int64_t)exp_avgs[i].data_ptr();
    h[3 * n + i] = (int64_t)exp_avg_sqs[i].data_ptr();
    h[4 * n + i] = params[i].numel();
    h[5 * n + i] = acc;
    acc += (params[i].numel() + kQBlock - 1) / kQBlock;
  }
  h[6 * n] = acc;
  at::Tensor dev = host.to(params[0].device(), /*non_blocking=*/true);
  const int64_t* d = dev.data_ptr<int64_t>();
  auto stream = at::cuda::getCurrentCUDAStream().stream();
  const float bc1 = 1.0f - std::pow((float)beta1, (float)step);
  const float bc2 = 1.0f - std::pow((float)beta2, (float)step);
  tft::launch_adamw(d, d + n, d + 2 * n, d + 3 * n, d + 4 * n, d + 5 * n, (int)n,
                    acc, (float)lr, (float)beta1, (float)beta2, (float)eps,
                    (float)weight_decay, bc1, bc2, (tft_stream)stream);
}

// ---- flash attention backward (appended) -----------------------------------

// true when a logical [B,H,S,D] tensor is a transpose view of contiguous
// [B,S,H,D] storage (the model's projection layout)
static bool is_bshd(const at::Tensor& t) {
  return t.dim() == 4 && t.stride(3) == 1 && t.stride(1) == t.size(3) &&
         t.stride(2) == t.size(1) * t.size(3) &&
         t.stride(0) == t.size(1) * t.size(2) * t.size(3);
}

// contiguous in logical [B,H,S,D] order
static bool is_bhsd(const at::Tensor& t) { return t.is_contiguous(); }

at::Tensor fa_delta(at::Tensor dout, at::Tensor out) {
  TORCH_CHECK(dout.is_cuda() && dout.scalar_type() == at::kBFloat16);
  TORCH_CHECK(dout.size(-1) == 128);
  const bool bshd = is_bshd(dout) && is_bshd(out);
  auto dc = bshd ? dout : dout.contiguous();
  auto oc = bshd ? out : out.contiguous();
  const int64_t rows = dc.numel() / 128;
  // delta is always dense [B, Hq, S]
  auto delta = at::empty({dout.size(0), dout.size(1), dout.size(2)},
                         dout.options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentCUDAStream().stream();
  tft::launch_fa_delta(dc.data_ptr(), oc.data_ptr(), delta.data_ptr<float>(),
                       rows, (int)dout.size(1), (int)dout.size(2), bshd,
                       (tft_stream)stream);
  return delta;
}

std::vector<at::Tensor> fa_fwd(at::Tensor q, at::Tensor k, at::Tensor v,
                               double scale, bool causal) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == at::kBFloat16 && q.dim() == 4);
  TORCH_CHECK(q.size(3) == 128, "fa_fwd supports head_dim=128 only");
  TORCH_CHECK(q.size(2) % 256 == 0, "fa_fwd requires seq % 256 == 0");
  const int64_t B = q.size(0), Hq = q.size(1), S = q.size(2);
  const int64_t Hkv = k.size(1);
  TORCH_CHECK(Hq % Hkv == 0, "Hq must be a multiple of Hkv");
  const bool bshd = is_bshd(q) && is_bshd(k) && is_bshd(v);
  auto qc = bshd ? q : q.contiguous();
  auto kc = bshd ? k : k.contiguous();
  auto vc = bshd ? v : v.contiguous();
  // out matches the input layout: with bshd storage the logical [B,Hq,S,D]
  // result is a transpose view of a [B,S,Hq,D] buffer (so a later
  // .transpose(1,2).reshape() in the model is free)
  auto out = bshd
                 ? at::empty({B, S, Hq, (int64_t)128}, q.options()).transpose(1, 2)
                 : at::empty_like(qc);
  auto lse = at::empty({B, Hq, S}, q.options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentCUDAStream().stream();
  tft::launch_fa_fwd(qc.data_ptr(), kc.data_ptr(), vc.data_ptr(),
                     out.data_ptr(), lse.data_ptr<float>(), (int)B, (int)Hq,
                     (int)Hkv, (int)S, (float)scale, causal, bshd,
                     (tft_stream)stream);
  return {out, lse};
}

std::vector<at::Tensor> fa_bwd(at::Tensor q, at::Tensor k, at::Tensor v,
                               at::Tensor dout, at::Tensor lse, at::Tensor delta,
                               double scale, bool causal) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == at::kBFloat16 && q.dim() == 4);
  TORCH_CHECK(q.size(3) == 128, "fa_bwd supports head_dim=128 only");
  TORCH_CHECK(q.size(2) % 128 == 0, "fa_bwd requires seq % 128 == 0");
  const int64_t B = q.size(0), Hq = q.size(1), S = q.size(2);
  const int64_t Hkv = k.size(1);
  TORCH_CHECK(Hq % Hkv == 0, "Hq must be a multiple of Hkv");
  const bool bshd = is_bshd(q) && is_bshd(k) && is_bshd(v) && is_bshd(dout);
  auto qc = bshd ? q : q.contiguous();
  auto kc = bshd ? k : k.contiguous();
  auto vc = bshd ? v : v.contiguous();
  auto doc = bshd ? dout : dout.contiguous();
  auto lsec = lse.contiguous().to(at::kFloat);
  auto deltac = delta.contiguous().to(at::kFloat);
  TORCH_CHECK(lsec.numel() == B * Hq * S, "lse shape mismatch");
  // gradients come back in the inputs' layout (bshd: transpose views of
  // [B,S,H,D] buffers) so the autograd graph above never copies
  auto mk = [&](const at::Tensor& like) {
    if (!bshd) return at::empty_like(like.contiguous());
    return at::empty({like.size(0), like.size(2), like.size(1), like.size(3)},
                     like.options())
        .transpose(1, 2);
  };
  auto dq = mk(q);
  auto dk = mk(k);
  auto dv = mk(v);
  auto stream = at::cuda::getCurrentCUDAStream().stream();
  tft::launch_fa_bwd(qc.data_ptr(), kc.data_ptr(), vc.data_ptr(), doc.data_ptr(),
                     lsec.data_ptr<float>(), deltac.data_ptr<float>(),
                     dq.data_ptr(), dk.data_ptr(), dv.data_ptr(), (int)B,
                     (int)Hq, (int)Hkv, (int)S, (float)scale, causal, bshd,
                     (tft_stream)stream);
  return {dq, dk, dv};
}

at::Tensor mfma_probe(at::Tensor A, at::Tensor B) {
  TORCH_CHECK(A.is_cuda() && A.scalar_type() == at::kBFloat16);
  TORCH_CHECK(A.sizes() == at::IntArrayRef({32, 16}) &&
              B.sizes() == at::IntArrayRef({16, 32}));
  auto D = at::zeros({32, 32}, A.options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentCUDAStream().stream();
  tft::launch_mfma_probe(A.contiguous().data_ptr(), B.contiguous().data_ptr(),
                         D.data_ptr<float>(), (tft_stream)stream);
  return D;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "torchft_amd CDNA4 (gfx950) HIP kernels";
  m.def("fp8_pack_bytes", &fp8_pack_bytes);
  m.def("fp8_slice_bytes", &fp8_slice_bytes);
  m.def("fp8_quantize", &fp8_quantize);
  m.def("fp8_dequantize", &fp8_dequantize);
  m.def("fp8_reduce", &fp8_reduce);
  m.def("rmsnorm_fwd", &rmsnorm_fwd);
  m.def("rmsnorm_bwd", &rmsnorm_bwd);
  m.def("rope_apply", &rope_apply);
  m.def("swiglu_fwd", &swiglu_fwd);
  m.def("swiglu_glu_fwd", &swiglu_glu_fwd);
  m.def("swiglu_glu_bwd", &swiglu_glu_bwd);
  m.def("swiglu_bwd", &swiglu_bwd);
  m.def("adamw_step", &adamw_step);
  m.def("fa_fwd", &fa_fwd);
  m.def("fa_delta", &fa_delta);
  m.def("fa_bwd", &fa_bwd,
        "flash-attention backward (bf16, D=128): returns (dq, dk, dv)");
  m.def("mfma_probe", &mfma_probe, "mfma_f32_32x32x16_bf16 layout probe");
}

