// Python bindings for the bdbnn_amd gfx950 kernels (pack / xnor conv /
// kurtosis / weight-KD / fused optimizers).  Pure HIP underneath — no CUDA
// compatibility layer; this file only marshals ATen tensors.
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

#include <string>
#include <unordered_map>
#include <vector>

// ---- mirrors of csrc/common.h (kept in sync) ----
constexpr int BDBNN_MAX_TENSORS = 64;
struct TensorListArg {
  const float* ptr[BDBNN_MAX_TENSORS];
  int64_t numel[BDBNN_MAX_TENSORS];
  int n;
};
struct PtrList { float* ptr[BDBNN_MAX_TENSORS]; };

extern "C" {
void bdbnn_sign_pack(const void*, uint32_t*, int64_t, int, int, bool,
                     hipStream_t);
void bdbnn_binsign_decode(const void*, void*, int64_t, bool, bool,
                          hipStream_t);
void bdbnn_ste_mask_mul(const void*, const void*, void*, int64_t, bool, bool,
                        int, float, float, hipStream_t);
void bdbnn_weight_pack(const float*, uint32_t*, float*, float*, int, int,
                       int, int, int, hipStream_t);
void bdbnn_xnor_conv_fwd(const uint32_t*, const uint32_t*, const float*,
                         const float*, void*, float*, float*, bool, int,
                         int, int, int, int, int, int, int, int, int, int,
                         hipStream_t);
void bdbnn_sign_mask_pack(const void*, uint32_t*, uint32_t*, int64_t, int,
                          int, bool, hipStream_t);
void bdbnn_decode_packed(const uint32_t*, void*, int64_t, int, int, bool,
                         hipStream_t);
void bdbnn_mask_mul_packed(const void*, const uint32_t*, void*, int64_t,
                           int, int, bool, bool, hipStream_t);
void bdbnn_weight_decode(const uint32_t*, const float*, void*, int, int,
                         int, int, bool, hipStream_t);
void bdbnn_maxpool_fwd(const void*, void*, unsigned char*, int, int, int,
                       int, int, int, int, int, int, bool, hipStream_t);
void bdbnn_maxpool_bwd(const void*, const unsigned char*, void*, int, int,
                       int, int, int, int, int, int, int, bool,
                       hipStream_t);
void bdbnn_kd_logit_fwd(const void*, const void*, float*, float*, int,
                        int, bool, hipStream_t);
void bdbnn_kd_logit_bwd(const void*, const void*, const float*, void*,
                        float, int, int, bool, hipStream_t);
void bdbnn_ce_fwd(const void*, const int64_t*, float*, float*, int, int,
                  bool, hipStream_t);
void bdbnn_ce_bwd(const void*, const int64_t*, const float*, void*, float,
                  int, int, bool, hipStream_t);
void bdbnn_conv_dgrad(const void*, const uint32_t*, const float*, void*,
                      int, int, int, int, int, int, hipStream_t);
int bdbnn_conv_dgrad2(const void*, const void*, const uint32_t*, void*,
                      const void*, int, int, int, int, int, hipStream_t);
void bdbnn_dgrad_wdec(const uint32_t*, const float*, void*, int, int,
                      hipStream_t);
int bdbnn_conv_wgrad2(const void*, const uint64_t*, float*, int, int, int,
                      int, int, hipStream_t);
int bdbnn_wgrad2_nslab(int, int, int, int, int);
void bdbnn_stem_pack_x4(const void*, void*, int64_t, bool, hipStream_t);
void bdbnn_stem_pack_w4(const float*, void*, hipStream_t);
void bdbnn_stem_fwd(const void*, const void*, void*, int, int, int,
                    hipStream_t);
int bdbnn_stem_wrw_nslab(int, int, int);
void bdbnn_stem_wrw(const void*, const void*, float*, int, int, int,
                    hipStream_t);
void bdbnn_stem_fold_dw4(const float*, float*, int, hipStream_t);
void bdbnn_repack_cplane(const uint32_t*, uint64_t*, int, int, int, int,
                         hipStream_t);
void bdbnn_wgrad_finish(const float*, const float*, float*, int, int, int,
                        hipStream_t);
void bdbnn_conv_wgrad(const void*, const uint32_t*, float*, int, int, int,
                      int, int, int, hipStream_t);
void bdbnn_prelu_fwd(const void*, const float*, void*, int64_t, int, bool,
                     hipStream_t);
void bdbnn_prelu_bwd(const void*, const void*, const float*, void*, float*,
                     int64_t, int, bool, hipStream_t);
void bdbnn_bn_stats(const void*, float*, float*, int64_t, int, bool,
                    hipStream_t);
void bdbnn_bn_finalize(const float*, const float*, float*, float*, float*,
                       float*, int, float, float, float, int, hipStream_t);
void bdbnn_bn_act_fwd(const void*, const void*, const float*, const float*,
                      const float*, const float*, const float*, void*, void*,
                      int64_t, int, int, uint32_t*, uint32_t*, bool,
                      hipStream_t);
void bdbnn_bn_act_bwd_reduce(const void*, const void*, const void*,
                             const float*, const float*, const float*,
                             float*, float*, int64_t, int, int, bool,
                             hipStream_t);
void bdbnn_bn_act_bwd_apply(const void*, const void*, const void*,
                            const float*, const float*, const float*,
                            const float*, const float*, void*, void*,
                            int64_t, int, int, float, bool, hipStream_t);
void bdbnn_bn_act_eval(const void*, const void*, const float*, const float*,
                       const float*, const float*, const float*, void*,
                       int64_t, int, int, float, bool, hipStream_t);
void bdbnn_kurtosis_fwd(const TensorListArg*, const int*, const int64_t*,
                        int, double*, const float*, float*, float*, float*,
                        hipStream_t);
void bdbnn_kurtosis_bwd(const TensorListArg*, const PtrList*, const int*,
                        const int64_t*, int, const float*, const float*,
                        const float*, hipStream_t);
void bdbnn_weight_kd_fwd(const TensorListArg*, const PtrList*, const PtrList*,
                         const int*, const int64_t*, int, double*,
                         hipStream_t);
void bdbnn_weight_kd_bwd(const TensorListArg*, const PtrList*, const PtrList*,
                         const int*, const int64_t*, int, const float*,
                         hipStream_t);
void bdbnn_fused_sgd(const TensorListArg*, const PtrList*, const PtrList*,
                     const PtrList*, const int*, const int64_t*, int, float,
                     float, float, hipStream_t);
void bdbnn_fused_adam(const TensorListArg*, const PtrList*, const PtrList*,
                      const PtrList*, const PtrList*, const int*,
                      const int64_t*, int, float, float, float, float, float,
                      float, float, hipStream_t);
}

namespace {

hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

bool is_bf16(const at::Tensor& t) { return t.scalar_type() == at::kBFloat16; }

// Build the multi-tensor block schedule (block -> tensor, offset) on the
// device.  chunk_elems must match the kernels' CHUNK_ELEMS.
struct Schedule {
  at::Tensor bt, bo;   // int32 / int64 device tensors
  int n_blocks;
};

Schedule build_schedule_uncached(const std::vector<at::Tensor>& ts,
                                 int64_t chunk_elems, const at::Device& dev) {
  std::vector<int> bt;
  std::vector<int64_t> bo;
  for (size_t l = 0; l < ts.size(); ++l) {
    int64_t n = ts[l].numel();
    for (int64_t off = 0; off < n; off += chunk_elems) {
      bt.push_back((int)l);
      bo.push_back(off);
    }
  }
  Schedule s;
  s.n_blocks = (int)bt.size();
  s.bt = at::from_blob(bt.data(), {(int64_t)bt.size()},
                       at::TensorOptions().dtype(at::kInt))
             .to(dev, /*non_blocking=*/false);
  s.bo = at::from_blob(bo.data(), {(int64_t)bo.size()},
                       at::TensorOptions().dtype(at::kLong))
             .to(dev, /*non_blocking=*/false);
  return s;
}

// The tensor sets of the fused calls are stable across steps (the same
// parameters every iteration): cache the device-side schedules keyed by
// the size list + chunking so the per-step host->device copies go away.
Schedule build_schedule(const std::vector<at::Tensor>& ts,
                        int64_t chunk_elems, const at::Device& dev) {
  static std::unordered_map<std::string, Schedule> cache;
  std::string key;
  key.reserve(ts.size() * 9 + 16);
  key += std::to_string(chunk_elems);
  key += '/';
  key += std::to_string(dev.index());
  for (auto& t : ts) {
    key += ':';
    key += std::to_string(t.numel());
  }
  auto it = cache.find(key);
  if (it != cache.end()) return it->second;
  if (cache.size() > 256) cache.clear();  // bound memory; rebuilt on demand
  auto s = build_schedule_uncached(ts, chunk_elems, dev);
  cache.emplace(std::move(key), s);
  return s;
}

TensorListArg make_meta(const std::vector<at::Tensor>& ts) {
  TORCH_CHECK((int)ts.size() <= BDBNN_MAX_TENSORS,
              "too many tensors for one fused call");
  TensorListArg a;
  a.n = (int)ts.size();
  for (size_t i = 0; i < ts.size(); ++i) {
    // dense layout in ANY memory format: the kernels are elementwise /
    // permutation-invariant over physical memory
    TORCH_CHECK((ts[i].is_contiguous() ||
                 ts[i].is_contiguous(at::MemoryFormat::ChannelsLast)) &&
                    ts[i].scalar_type() == at::kFloat,
                "fused multi-tensor ops need dense fp32");
    a.ptr[i] = ts[i].data_ptr<float>();
    a.numel[i] = ts[i].numel();
  }
  return a;
}

PtrList make_ptrs(const std::vector<at::Tensor>& ts) {
  PtrList p;
  for (size_t i = 0; i < ts.size(); ++i) p.ptr[i] = ts[i].data_ptr<float>();
  return p;
}

}  // namespace

// ---------------- pack / quantizer ----------------

at::Tensor sign_pack_nhwc(const at::Tensor& x) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4, "sign_pack: 4-D CUDA tensor");
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast),
              "sign_pack: channels_last input required");
  int64_t N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  int CW = (int)((C + 31) / 32);
  auto out = at::empty({N, H, W, CW},
                       x.options().dtype(at::kInt));
  bdbnn_sign_pack(x.data_ptr(), (uint32_t*)out.data_ptr<int>(),
                  N * H * W, (int)C, CW, is_bf16(x), cur_stream());
  return out;
}

std::vector<at::Tensor> weight_pack(const at::Tensor& w) {
  TORCH_CHECK(w.is_cuda() && w.dim() == 4, "weight_pack: 4-D CUDA tensor");
  auto wf = w.contiguous().to(at::kFloat);
  int K = (int)wf.size(0), C = (int)wf.size(1);
  int KH = (int)wf.size(2), KW = (int)wf.size(3);
  int CW = (C + 31) / 32;
  auto wp = at::empty({K, KH, KW, CW}, wf.options().dtype(at::kInt));
  auto alpha = at::empty({K}, wf.options());
  auto stab = at::empty({K, KH * KW}, wf.options());
  bdbnn_weight_pack(wf.data_ptr<float>(), (uint32_t*)wp.data_ptr<int>(),
                    alpha.data_ptr<float>(), stab.data_ptr<float>(), K, C,
                    KH, KW, CW, cur_stream());
  return {wp, alpha, stab};
}

at::Tensor binsign_decode(const at::Tensor& x, bool out_bf16) {
  TORCH_CHECK(x.is_cuda(), "binsign_decode: CUDA tensor");
  TORCH_CHECK(x.numel() % 8 == 0, "binsign_decode: numel % 8 == 0");
  auto fmt = x.dim() == 4 ? at::MemoryFormat::ChannelsLast
                          : at::MemoryFormat::Contiguous;
  auto xc = x.contiguous(fmt);
  auto out = at::empty_like(xc, xc.options().dtype(
      out_bf16 ? at::kBFloat16 : at::kFloat), fmt);
  bdbnn_binsign_decode(xc.data_ptr(), out.data_ptr(), xc.numel(),
                       is_bf16(xc), out_bf16, cur_stream());
  return out;
}

at::Tensor ste_mask_mul(const at::Tensor& g, const at::Tensor& x, int mode,
                        double t, double k) {
  TORCH_CHECK(g.is_cuda() && x.is_cuda() && g.numel() == x.numel(),
              "ste_mask_mul: matching CUDA tensors");
  TORCH_CHECK(x.numel() % 8 == 0, "ste_mask_mul: numel % 8 == 0");
  auto fmt = x.dim() == 4 ? at::MemoryFormat::ChannelsLast
                          : at::MemoryFormat::Contiguous;
  auto xc = x.contiguous(fmt);
  auto gc = g.contiguous(fmt);
  auto out = at::empty_like(xc, xc.options(), fmt);
  bdbnn_ste_mask_mul(gc.data_ptr(), xc.data_ptr(), out.data_ptr(),
                     xc.numel(), is_bf16(gc), is_bf16(xc), mode, (float)t,
                     (float)k, cur_stream());
  return out;
}

// ---------------- packed-bit fast path ----------------

std::vector<at::Tensor> sign_mask_pack_nhwc(const at::Tensor& x) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4, "sign_mask_pack: 4-D CUDA tensor");
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast),
              "sign_mask_pack: channels_last input required");
  int64_t N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  int CW = (int)((C + 31) / 32);
  auto sp = at::empty({N, H, W, CW}, x.options().dtype(at::kInt));
  auto mp = at::empty({N, H, W, CW}, x.options().dtype(at::kInt));
  bdbnn_sign_mask_pack(x.data_ptr(), (uint32_t*)sp.data_ptr<int>(),
                       (uint32_t*)mp.data_ptr<int>(), N * H * W, (int)C, CW,
                       is_bf16(x), cur_stream());
  return {sp, mp};
}

at::Tensor decode_packed(const at::Tensor& sp, int64_t C, bool out_bf16) {
  TORCH_CHECK(sp.is_cuda() && sp.dim() == 4 && sp.scalar_type() == at::kInt,
              "decode_packed: int32 [N,H,W,CW]");
  int64_t N = sp.size(0), H = sp.size(1), W = sp.size(2);
  int CW = (int)sp.size(3);
  auto out = at::empty({N, C, H, W},
                       sp.options().dtype(out_bf16 ? at::kBFloat16
                                                   : at::kFloat),
                       at::MemoryFormat::ChannelsLast);
  bdbnn_decode_packed((const uint32_t*)sp.data_ptr<int>(), out.data_ptr(),
                      N * H * W, (int)C, CW, out_bf16, cur_stream());
  return out;
}

at::Tensor mask_mul_packed(const at::Tensor& g, const at::Tensor& mp,
                           int64_t C, bool out_bf16) {
  TORCH_CHECK(g.is_cuda() && mp.is_cuda(), "mask_mul_packed: CUDA tensors");
  auto gc = g.contiguous(at::MemoryFormat::ChannelsLast);
  int64_t N = mp.size(0), H = mp.size(1), W = mp.size(2);
  int CW = (int)mp.size(3);
  TORCH_CHECK(gc.numel() == N * H * W * C, "mask_mul_packed: shape mismatch");
  auto out = at::empty({N, C, H, W},
                       g.options().dtype(out_bf16 ? at::kBFloat16
                                                  : at::kFloat),
                       at::MemoryFormat::ChannelsLast);
  bdbnn_mask_mul_packed(gc.data_ptr(), (const uint32_t*)mp.data_ptr<int>(),
                        out.data_ptr(), N * H * W, (int)C, CW, is_bf16(gc),
                        out_bf16, cur_stream());
  return out;
}

at::Tensor weight_decode(const at::Tensor& wp, const at::Tensor& alpha,
                         int64_t C, bool out_bf16) {
  TORCH_CHECK(wp.is_cuda() && wp.dim() == 4 && wp.scalar_type() == at::kInt,
              "weight_decode: int32 [K,KH,KW,CW]");
  int K = (int)wp.size(0), KH = (int)wp.size(1), KW = (int)wp.size(2);
  int CW = (int)wp.size(3);
  auto out = at::empty({K, C, KH, KW},
                       wp.options().dtype(out_bf16 ? at::kBFloat16
                                                   : at::kFloat));
  bdbnn_weight_decode((const uint32_t*)wp.data_ptr<int>(),
                      alpha.data_ptr<float>(), out.data_ptr(), K, (int)C,
                      KH * KW, CW, out_bf16, cur_stream());
  return out;
}

// ---------------- xnor conv ----------------

std::vector<at::Tensor> xnor_conv_fwd(
    const at::Tensor& xp, const at::Tensor& wp, const at::Tensor& alpha,
    const at::Tensor& stab, int64_t C, int64_t stride, int64_t pad,
    bool out_bf16, bool want_stats) {
  TORCH_CHECK(xp.is_cuda() && xp.dim() == 4 && xp.scalar_type() == at::kInt,
              "xnor_conv: packed activations int32 [N,H,W,CW]");
  TORCH_CHECK(wp.dim() == 4, "xnor_conv: packed weights [K,KH,KW,CW]");
  int N = (int)xp.size(0), H = (int)xp.size(1), W = (int)xp.size(2);
  int K = (int)wp.size(0), KH = (int)wp.size(1), KW = (int)wp.size(2);
  TORCH_CHECK(KH * KW * wp.size(3) <= 160,
              "xnor_conv: KH*KW*ceil(C/32) must be <= 160 "
              "(kernel address-table size; C <= 512 at 3x3)");
  int Ho = (int)((H + 2 * pad - KH) / stride + 1);
  int Wo = (int)((W + 2 * pad - KW) / stride + 1);
  auto out = at::empty({N, K, Ho, Wo},
                       xp.options().dtype(out_bf16 ? at::kBFloat16
                                                   : at::kFloat),
                       at::MemoryFormat::ChannelsLast);
  at::Tensor s1, s2;
  float *s1p = nullptr, *s2p = nullptr;
  if (want_stats) {
    auto fopt = xp.options().dtype(at::kFloat);
    s1 = at::empty({32, K}, fopt);   // 32-way sliced partial sums
    s2 = at::empty({32, K}, fopt);
    s1p = s1.data_ptr<float>();
    s2p = s2.data_ptr<float>();
  }
  bdbnn_xnor_conv_fwd((const uint32_t*)xp.data_ptr<int>(),
                      (const uint32_t*)wp.data_ptr<int>(),
                      alpha.data_ptr<float>(), stab.data_ptr<float>(),
                      out.data_ptr(), s1p, s2p, out_bf16, N, H, W, (int)C,
                      K, KH, KW, (int)stride, (int)pad, Ho, Wo,
                      cur_stream());
  if (want_stats) return {out, s1, s2};
  return {out};
}


// ---------------- stem conv (7x7/2, C=3 -> 64) ----------------

std::vector<at::Tensor> stem_conv_fwd(const at::Tensor& x,
                                      const at::Tensor& w) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 && x.size(1) == 3,
              "stem: (N,3,H,W) input");
  TORCH_CHECK(w.dim() == 4 && w.size(0) == 64 && w.size(1) == 3 &&
                  w.size(2) == 7 && w.size(3) == 7,
              "stem: (64,3,7,7) weights");
  int N = (int)x.size(0), H = (int)x.size(2), W = (int)x.size(3);
  TORCH_CHECK(H % 2 == 0 && W % 2 == 0, "stem: even H,W");
  auto xc = x.contiguous(at::MemoryFormat::ChannelsLast);
  bool bf16 = is_bf16(xc);
  auto bopt = xc.options().dtype(at::kBFloat16);
  auto x4 = at::empty({N, H, W, 4}, bopt);
  bdbnn_stem_pack_x4(xc.data_ptr(), x4.data_ptr(), (int64_t)N * H * W,
                     bf16, cur_stream());
  auto wf = w.contiguous().to(at::kFloat);
  auto w4T = at::empty({64, 224}, bopt);
  bdbnn_stem_pack_w4(wf.data_ptr<float>(), w4T.data_ptr(), cur_stream());
  auto out = at::empty({N, 64, H / 2, W / 2}, bopt,
                       at::MemoryFormat::ChannelsLast);
  bdbnn_stem_fwd(x4.data_ptr(), w4T.data_ptr(), out.data_ptr(), N, H, W,
                 cur_stream());
  return {out, x4};
}

at::Tensor stem_conv_wrw(const at::Tensor& x4, const at::Tensor& gy) {
  TORCH_CHECK(x4.is_cuda() && x4.dim() == 4 && x4.size(3) == 4 &&
                  x4.scalar_type() == at::kBFloat16,
              "stem wrw: packed x4 (N,H,W,4) bf16");
  int N = (int)x4.size(0), H = (int)x4.size(1), W = (int)x4.size(2);
  auto gc = gy.to(at::kBFloat16).contiguous(at::MemoryFormat::ChannelsLast);
  TORCH_CHECK(gc.size(1) == 64 && gc.size(2) == H / 2 &&
                  gc.size(3) == W / 2, "stem wrw: grad shape");
  int nslab = bdbnn_stem_wrw_nslab(N, H, W);
  auto fopt = x4.options().dtype(at::kFloat);
  auto slabs = at::empty({nslab, 64, 224}, fopt);
  bdbnn_stem_wrw(x4.data_ptr(), gc.data_ptr(), slabs.data_ptr<float>(),
                 N, H, W, cur_stream());
  auto dw = at::empty({64, 3, 7, 7}, fopt);
  bdbnn_stem_fold_dw4(slabs.data_ptr<float>(), dw.data_ptr<float>(),
                      nslab, cur_stream());
  return dw;
}

// ---------------- fused classification losses ----------------

std::vector<at::Tensor> kd_logit_fwd(const at::Tensor& s,
                                     const at::Tensor& t) {
  TORCH_CHECK(s.is_cuda() && s.dim() == 2 && s.sizes() == t.sizes(),
              "kd_logit: (B,C) logits");
  auto sc = s.contiguous();
  auto tc = t.to(s.scalar_type()).contiguous();
  int B = (int)s.size(0), C = (int)s.size(1);
  auto fopt = sc.options().dtype(at::kFloat);
  auto stats = at::empty({B, 4}, fopt);
  auto out = at::empty({}, fopt);
  bdbnn_kd_logit_fwd(sc.data_ptr(), tc.data_ptr(), stats.data_ptr<float>(),
                     out.data_ptr<float>(), B, C, is_bf16(sc),
                     cur_stream());
  return {out, stats, tc};
}

at::Tensor kd_logit_bwd(const at::Tensor& s, const at::Tensor& t,
                        const at::Tensor& stats, double gscale) {
  auto sc = s.contiguous();
  int B = (int)s.size(0), C = (int)s.size(1);
  auto ds = at::empty_like(sc);
  bdbnn_kd_logit_bwd(sc.data_ptr(), t.data_ptr(), stats.data_ptr<float>(),
                     ds.data_ptr(), (float)(gscale / B), B, C, is_bf16(sc),
                     cur_stream());
  return ds;
}

std::vector<at::Tensor> ce_fwd(const at::Tensor& s, const at::Tensor& y) {
  TORCH_CHECK(s.is_cuda() && s.dim() == 2 && y.scalar_type() == at::kLong,
              "ce: (B,C) logits + int64 targets");
  auto sc = s.contiguous();
  auto yc = y.contiguous();
  int B = (int)s.size(0), C = (int)s.size(1);
  auto fopt = sc.options().dtype(at::kFloat);
  auto stats = at::empty({B}, fopt);
  auto out = at::empty({}, fopt);
  bdbnn_ce_fwd(sc.data_ptr(), yc.data_ptr<int64_t>(),
               stats.data_ptr<float>(), out.data_ptr<float>(), B, C,
               is_bf16(sc), cur_stream());
  return {out, stats};
}

at::Tensor ce_bwd(const at::Tensor& s, const at::Tensor& y,
                  const at::Tensor& stats, double gscale) {
  auto sc = s.contiguous();
  int B = (int)s.size(0), C = (int)s.size(1);
  auto ds = at::empty_like(sc);
  bdbnn_ce_bwd(sc.data_ptr(), y.data_ptr<int64_t>(),
               stats.data_ptr<float>(), ds.data_ptr(),
               (float)(gscale / B), B, C, is_bf16(sc), cur_stream());
  return ds;
}

// ---------------- maxpool ----------------

std::vector<at::Tensor> maxpool_fwd(const at::Tensor& x, int64_t ks,
                                    int64_t stride, int64_t pad) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4, "maxpool_fwd: 4-D CUDA tensor");
  auto xc = x.contiguous(at::MemoryFormat::ChannelsLast);
  int N = (int)x.size(0), C = (int)x.size(1);
  int H = (int)x.size(2), W = (int)x.size(3);
  int Ho = (int)((H + 2 * pad - ks) / stride + 1);
  int Wo = (int)((W + 2 * pad - ks) / stride + 1);
  auto out = at::empty({N, C, Ho, Wo}, xc.options(),
                       at::MemoryFormat::ChannelsLast);
  auto idx = at::empty({N, Ho, Wo, C}, xc.options().dtype(at::kByte));
  bdbnn_maxpool_fwd(xc.data_ptr(), out.data_ptr(),
                    idx.data_ptr<unsigned char>(), N, C, H, W, Ho, Wo,
                    (int)ks, (int)stride, (int)pad, is_bf16(xc),
                    cur_stream());
  return {out, idx};
}

at::Tensor maxpool_bwd(const at::Tensor& dy, const at::Tensor& idx,
                       int64_t H, int64_t W, int64_t ks, int64_t stride,
                       int64_t pad) {
  auto dyc = dy.contiguous(at::MemoryFormat::ChannelsLast);
  int N = (int)dy.size(0), C = (int)dy.size(1);
  int Ho = (int)dy.size(2), Wo = (int)dy.size(3);
  auto dx = at::empty({N, C, H, W}, dyc.options(),
                      at::MemoryFormat::ChannelsLast);
  bdbnn_maxpool_bwd(dyc.data_ptr(), idx.data_ptr<unsigned char>(),
                    dx.data_ptr(), N, C, (int)H, (int)W, Ho, Wo, (int)ks,
                    (int)stride, (int)pad, is_bf16(dyc), cur_stream());
  return dx;
}

// ---------------- prelu ----------------

at::Tensor prelu_fwd(const at::Tensor& x, const at::Tensor& a) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4, "prelu_fwd: 4-D CUDA tensor");
  auto xc = x.contiguous(at::MemoryFormat::ChannelsLast);
  auto af = a.contiguous().to(at::kFloat);
  int C = (int)x.size(1);
  TORCH_CHECK(C <= 1024, "prelu: C <= 1024");
  auto y = at::empty_like(xc, xc.options(), at::MemoryFormat::ChannelsLast);
  bdbnn_prelu_fwd(xc.data_ptr(), af.data_ptr<float>(), y.data_ptr(),
                  xc.numel(), C, xc.scalar_type() == at::kBFloat16,
                  cur_stream());
  return y;
}

std::vector<at::Tensor> prelu_bwd(const at::Tensor& g, const at::Tensor& x,
                                  const at::Tensor& a) {
  auto xc = x.contiguous(at::MemoryFormat::ChannelsLast);
  auto gc = g.contiguous(at::MemoryFormat::ChannelsLast);
  TORCH_CHECK(gc.scalar_type() == xc.scalar_type(),
              "prelu_bwd: grad dtype must match input");
  auto af = a.contiguous().to(at::kFloat);
  int C = (int)x.size(1);
  auto dx = at::empty_like(xc, xc.options(), at::MemoryFormat::ChannelsLast);
  auto da = at::empty({C}, xc.options().dtype(at::kFloat));
  bdbnn_prelu_bwd(xc.data_ptr(), gc.data_ptr(), af.data_ptr<float>(),
                  dx.data_ptr(), da.data_ptr<float>(), xc.numel(), C,
                  xc.scalar_type() == at::kBFloat16, cur_stream());
  return {dx, da};
}

// ---------------- fused BN (+add) (+act) ----------------

std::vector<at::Tensor> bn_act_fwd_train(
    const at::Tensor& x, const c10::optional<at::Tensor>& skip,
    const at::Tensor& gamma, const at::Tensor& beta,
    const c10::optional<at::Tensor>& a,
    c10::optional<at::Tensor> running_mean,
    c10::optional<at::Tensor> running_var, double momentum, double eps,
    int64_t act_kind, const c10::optional<at::Tensor>& pre_s1,
    const c10::optional<at::Tensor>& pre_s2, bool want_pack) {
  auto xc = x.contiguous(at::MemoryFormat::ChannelsLast);
  int C = (int)x.size(1);
  TORCH_CHECK(C <= 1024, "fused bn: C <= 1024");
  int64_t n = xc.numel();
  bool bf16 = is_bf16(xc);
  auto fopt = xc.options().dtype(at::kFloat);
  at::Tensor s1, s2;
  int nslice = 1;
  auto mean = at::empty({C}, fopt);
  auto invstd = at::empty({C}, fopt);
  if (pre_s1.has_value()) {
    // stats already accumulated by the producing conv's epilogue
    // (possibly as [nslice][C] partial sums — finalize folds them)
    s1 = *pre_s1;
    s2 = *pre_s2;
    nslice = (int)(s1.numel() / C);
    TORCH_CHECK(nslice * C == s1.numel(), "bn: stats shape mismatch");
  } else {
    s1 = at::empty({32, C}, fopt);   // sliced partials (finalize folds)
    s2 = at::empty({32, C}, fopt);
    nslice = 32;
    bdbnn_bn_stats(xc.data_ptr(), s1.data_ptr<float>(),
                   s2.data_ptr<float>(), n, C, bf16, cur_stream());
  }
  float* rm = running_mean.has_value()
                  ? running_mean->data_ptr<float>() : nullptr;
  float* rv = running_var.has_value()
                  ? running_var->data_ptr<float>() : nullptr;
  bdbnn_bn_finalize(s1.data_ptr<float>(), s2.data_ptr<float>(),
                    mean.data_ptr<float>(), invstd.data_ptr<float>(), rm, rv,
                    C, (float)(n / C), (float)momentum, (float)eps, nslice,
                    cur_stream());
  auto out = at::empty_like(xc, xc.options(), at::MemoryFormat::ChannelsLast);
  // PReLU backward needs the true pre-activation z; ReLU only needs its
  // sign (recoverable from out) and identity ignores it — skip the write.
  at::Tensor z;
  void* zptr = nullptr;
  if (act_kind == 1) {
    z = at::empty_like(xc, xc.options(), at::MemoryFormat::ChannelsLast);
    zptr = z.data_ptr();
  } else {
    z = out;
  }
  at::Tensor skipc;
  const void* skip_ptr = nullptr;
  if (skip.has_value()) {
    skipc = skip->to(xc.scalar_type())
                .contiguous(at::MemoryFormat::ChannelsLast);
    skip_ptr = skipc.data_ptr();
  }
  auto gf = gamma.contiguous().to(at::kFloat);
  auto bf = beta.contiguous().to(at::kFloat);
  at::Tensor af;
  const float* a_ptr = nullptr;
  if (a.has_value()) { af = a->contiguous().to(at::kFloat);
                       a_ptr = af.data_ptr<float>(); }
  // consumer-conv sign/mask pack fused into the epilogue (the next
  // binary conv then skips its own pack read pass; csrc/bn_act.hip)
  at::Tensor xpk, mpk;
  uint32_t *xpk_p = nullptr, *mpk_p = nullptr;
  if (want_pack) {
    TORCH_CHECK(C % 32 == 0, "bn pack fusion needs C % 32 == 0");
    int CW = C / 32;
    auto iopt = xc.options().dtype(at::kInt);
    xpk = at::empty({x.size(0), x.size(2), x.size(3), CW}, iopt);
    mpk = at::empty({x.size(0), x.size(2), x.size(3), CW}, iopt);
    xpk_p = (uint32_t*)xpk.data_ptr<int>();
    mpk_p = (uint32_t*)mpk.data_ptr<int>();
  }
  bdbnn_bn_act_fwd(xc.data_ptr(), skip_ptr, mean.data_ptr<float>(),
                   invstd.data_ptr<float>(), gf.data_ptr<float>(),
                   bf.data_ptr<float>(), a_ptr, out.data_ptr(), zptr,
                   n, C, (int)act_kind, xpk_p, mpk_p, bf16, cur_stream());
  if (want_pack) return {out, z, mean, invstd, xpk, mpk};
  return {out, z, mean, invstd};
}

std::vector<at::Tensor> bn_act_bwd(
    const at::Tensor& dy, const at::Tensor& z, const at::Tensor& x,
    const at::Tensor& mean, const at::Tensor& invstd,
    const at::Tensor& gamma, const c10::optional<at::Tensor>& a,
    int64_t act_kind, bool need_dskip) {
  auto xc = x.contiguous(at::MemoryFormat::ChannelsLast);
  auto zc = z.contiguous(at::MemoryFormat::ChannelsLast);
  auto dyc = dy.to(xc.scalar_type()).contiguous(at::MemoryFormat::ChannelsLast);
  int C = (int)x.size(1);
  int64_t n = xc.numel();
  bool bf16 = is_bf16(xc);
  auto fopt = xc.options().dtype(at::kFloat);
  auto sums32 = at::empty({32, C, 3}, fopt);  // sliced partials
  auto sums = at::empty({C, 3}, fopt);
  at::Tensor af;
  const float* a_ptr = nullptr;
  if (a.has_value()) { af = a->contiguous().to(at::kFloat);
                       a_ptr = af.data_ptr<float>(); }
  bdbnn_bn_act_bwd_reduce(dyc.data_ptr(), zc.data_ptr(), xc.data_ptr(),
                          mean.data_ptr<float>(), invstd.data_ptr<float>(),
                          a_ptr, sums32.data_ptr<float>(),
                          sums.data_ptr<float>(), n, C, (int)act_kind,
                          bf16, cur_stream());
  auto gf = gamma.contiguous().to(at::kFloat);
  auto dx = at::empty_like(xc, xc.options(), at::MemoryFormat::ChannelsLast);
  at::Tensor dskip;
  void* dskip_ptr = nullptr;
  if (need_dskip) {
    dskip = at::empty_like(xc, xc.options(), at::MemoryFormat::ChannelsLast);
    dskip_ptr = dskip.data_ptr();
  } else {
    dskip = at::empty({0}, xc.options());
  }
  bdbnn_bn_act_bwd_apply(dyc.data_ptr(), zc.data_ptr(), xc.data_ptr(),
                         mean.data_ptr<float>(), invstd.data_ptr<float>(),
                         gf.data_ptr<float>(), a_ptr, sums.data_ptr<float>(),
                         dx.data_ptr(), dskip_ptr, n, C, (int)act_kind,
                         (float)(1.0 / (double)(n / C)), bf16, cur_stream());
  auto dbeta = sums.select(1, 0).clone();
  auto dgamma = sums.select(1, 1).clone();
  auto da = sums.select(1, 2).clone();
  return {dx, dskip, dgamma, dbeta, da};
}

at::Tensor bn_act_eval(const at::Tensor& x,
                       const c10::optional<at::Tensor>& skip,
                       const at::Tensor& gamma, const at::Tensor& beta,
                       const c10::optional<at::Tensor>& a,
                       const at::Tensor& rm, const at::Tensor& rv,
                       double eps, int64_t act_kind) {
  auto xc = x.contiguous(at::MemoryFormat::ChannelsLast);
  int C = (int)x.size(1);
  int64_t n = xc.numel();
  bool bf16 = is_bf16(xc);
  auto out = at::empty_like(xc, xc.options(), at::MemoryFormat::ChannelsLast);
  at::Tensor skipc;
  const void* skip_ptr = nullptr;
  if (skip.has_value()) {
    skipc = skip->to(xc.scalar_type())
                .contiguous(at::MemoryFormat::ChannelsLast);
    skip_ptr = skipc.data_ptr();
  }
  auto gf = gamma.contiguous().to(at::kFloat);
  auto bf = beta.contiguous().to(at::kFloat);
  auto rmf = rm.contiguous().to(at::kFloat);
  auto rvf = rv.contiguous().to(at::kFloat);
  at::Tensor af;
  const float* a_ptr = nullptr;
  if (a.has_value()) { af = a->contiguous().to(at::kFloat);
                       a_ptr = af.data_ptr<float>(); }
  bdbnn_bn_act_eval(xc.data_ptr(), skip_ptr, rmf.data_ptr<float>(),
                    rvf.data_ptr<float>(), gf.data_ptr<float>(),
                    bf.data_ptr<float>(), a_ptr, out.data_ptr(), n, C,
                    (int)act_kind, (float)eps, bf16, cur_stream());
  return out;
}

// ---------------- MFMA dgrad v2 (hot path) ----------------

at::Tensor dgrad_weight_decode(const at::Tensor& wp, const at::Tensor& alpha,
                               int64_t C) {
  TORCH_CHECK(wp.is_cuda() && wp.dim() == 4 && wp.size(1) == 3 &&
                  wp.size(2) == 3,
              "dgrad_weight_decode: packed 3x3 weights [K][3][3][CW]");
  int K = (int)wp.size(0);
  TORCH_CHECK(K % 8 == 0 && C % 32 == 0, "dgrad_weight_decode: K%8, C%32");
  auto wd = at::empty({9, C, K}, wp.options().dtype(at::kBFloat16));
  auto al = alpha.contiguous();
  auto wpc = wp.contiguous();
  bdbnn_dgrad_wdec((const uint32_t*)wpc.data_ptr<int>(),
                   al.data_ptr<float>(), wd.data_ptr(), (int)C, K,
                   cur_stream());
  return wd;
}

// supported(N,H,W,C,K) mirror of the kernel's launch table
static bool dgrad2_ok(int H, int W, int C, int K) {
  if (C % 64 || K % 64 || W > 64) return false;
  if (W <= 8 && H > 8) return false;
  return true;
}

at::Tensor conv_dgrad2(const at::Tensor& g, const at::Tensor& wd,
                       const at::Tensor& mp, int64_t C,
                       const c10::optional<at::Tensor>& acc) {
  TORCH_CHECK(g.is_cuda() && g.dim() == 4 &&
                  g.scalar_type() == at::kBFloat16 &&
                  g.is_contiguous(at::MemoryFormat::ChannelsLast),
              "conv_dgrad2: bf16 channels_last grad");
  int N = (int)g.size(0), K = (int)g.size(1);
  int H = (int)g.size(2), W = (int)g.size(3);
  TORCH_CHECK(wd.dim() == 3 && wd.size(0) == 9 && wd.size(1) == C &&
                  wd.size(2) == K && wd.scalar_type() == at::kBFloat16,
              "conv_dgrad2: decoded weights [9][C][K] bf16");
  TORCH_CHECK(mp.size(-1) == C / 32 &&
                  mp.numel() == (int64_t)N * H * W * (C / 32),
              "conv_dgrad2: mask bitplane [...P...][C/32]");
  TORCH_CHECK(dgrad2_ok(H, W, (int)C, K), "conv_dgrad2: unsupported shape");
  const void* accp = nullptr;
  at::Tensor ac;
  if (acc.has_value()) {
    ac = acc->scalar_type() == at::kBFloat16
             ? acc->contiguous(at::MemoryFormat::ChannelsLast)
             : acc->to(at::kBFloat16)
                   .contiguous(at::MemoryFormat::ChannelsLast);
    TORCH_CHECK(ac.dim() == 4 && ac.size(0) == N && ac.size(1) == C &&
                    ac.size(2) == H && ac.size(3) == W,
                "conv_dgrad2: acc shape must match dx (N,C,H,W)");
    accp = ac.data_ptr();
  }
  auto dx = at::empty({N, C, H, W}, g.options(),
                      at::MemoryFormat::ChannelsLast);
  int rc = bdbnn_conv_dgrad2(g.data_ptr(), wd.data_ptr(),
                             (const uint32_t*)mp.data_ptr<int>(),
                             dx.data_ptr(), accp, N, H, W, (int)C, K,
                             cur_stream());
  TORCH_CHECK(rc == 0, "conv_dgrad2: launch rejected the shape");
  return dx;
}

// ---------------- MFMA wgrad v2 (hot path) ----------------

at::Tensor repack_cplane(const at::Tensor& xp, int64_t C, int64_t W) {
  TORCH_CHECK(xp.is_cuda() && xp.dim() == 4 && xp.size(3) == (C + 31) / 32,
              "repack_cplane: packed activations [N][H][W][CW]");
  TORCH_CHECK(xp.size(2) == W, "repack_cplane: W mismatch");
  TORCH_CHECK(W <= 64, "repack_cplane: W <= 64");
  int NH = (int)(xp.size(0) * xp.size(1));
  auto xcp = at::empty({C, NH}, xp.options().dtype(at::kLong));
  bdbnn_repack_cplane((const uint32_t*)xp.data_ptr<int>(),
                      (uint64_t*)xcp.data_ptr<int64_t>(), NH, (int)W,
                      (int)C, (int)xp.size(3), cur_stream());
  return xcp;
}

at::Tensor conv_wgrad2(const at::Tensor& g, const at::Tensor& xcp,
                       int64_t C) {
  TORCH_CHECK(g.is_cuda() && g.dim() == 4 &&
                  g.scalar_type() == at::kBFloat16 &&
                  g.is_contiguous(at::MemoryFormat::ChannelsLast),
              "conv_wgrad2: bf16 channels_last grad");
  int N = (int)g.size(0), K = (int)g.size(1);
  int H = (int)g.size(2), W = (int)g.size(3);
  TORCH_CHECK(xcp.dim() == 2 && xcp.size(0) == C &&
                  xcp.size(1) == (int64_t)N * H,
              "conv_wgrad2: c-plane bitplanes [C][N*H]");
  TORCH_CHECK(C % 64 == 0 && K % 64 == 0 && W <= 64,
              "conv_wgrad2: unsupported shape");
  int nslab = bdbnn_wgrad2_nslab(N, H, W, (int)C, K);
  TORCH_CHECK(nslab > 0, "conv_wgrad2: unsupported shape");
  // per-m-split-block partial slabs; every word is written (no memset)
  auto dwT = at::empty({nslab, 9, C, K}, g.options().dtype(at::kFloat));
  int rc = bdbnn_conv_wgrad2(g.data_ptr(),
                             (const uint64_t*)xcp.data_ptr<int64_t>(),
                             dwT.data_ptr<float>(), N, H, W, (int)C, K,
                             cur_stream());
  TORCH_CHECK(rc == 0, "conv_wgrad2: launch rejected the shape");
  return dwT;
}

at::Tensor wgrad_finish(const at::Tensor& dwT, const at::Tensor& w) {
  TORCH_CHECK(w.is_cuda() && w.dim() == 4 && w.size(2) == 3 &&
                  w.size(3) == 3 && w.scalar_type() == at::kFloat,
              "wgrad_finish: fp32 [K][C][3][3] latent weights");
  int K = (int)w.size(0), C = (int)w.size(1);
  TORCH_CHECK(dwT.dim() == 4 && dwT.size(1) == 9 && dwT.size(2) == C &&
                  dwT.size(3) == K && dwT.scalar_type() == at::kFloat,
              "wgrad_finish: dwT [nslab][9][C][K] fp32");
  auto wc = w.contiguous();
  auto dw = at::empty_like(wc);
  bdbnn_wgrad_finish(dwT.contiguous().data_ptr<float>(),
                     wc.data_ptr<float>(), dw.data_ptr<float>(), C, K,
                     (int)dwT.size(0), cur_stream());
  return dw;
}

// ---------------- experimental MFMA dgrad (v1, kept for A/B) -------------

at::Tensor conv_dgrad(const at::Tensor& g, const at::Tensor& wp,
                      const at::Tensor& alpha, int64_t C) {
  TORCH_CHECK(g.is_cuda() && g.dim() == 4 &&
                  g.scalar_type() == at::kBFloat16 &&
                  g.is_contiguous(at::MemoryFormat::ChannelsLast),
              "conv_dgrad: bf16 channels_last grad");
  int N = (int)g.size(0), K = (int)g.size(1);
  int H = (int)g.size(2), W = (int)g.size(3);
  TORCH_CHECK(wp.size(0) == K && wp.size(1) == 3 && wp.size(2) == 3,
              "conv_dgrad: packed 3x3 weights");
  TORCH_CHECK(((int64_t)N * H * W) % 128 == 0 && C % 64 == 0 && K % 16 == 0,
              "conv_dgrad: shape constraints");
  int CW = (int)wp.size(3);
  auto dx = at::empty({N, C, H, W}, g.options(),
                      at::MemoryFormat::ChannelsLast);
  bdbnn_conv_dgrad(g.data_ptr(), (const uint32_t*)wp.data_ptr<int>(),
                   alpha.data_ptr<float>(), dx.data_ptr(), N, H, W, (int)C,
                   K, CW, cur_stream());
  return dx;
}

at::Tensor conv_wgrad(const at::Tensor& g, const at::Tensor& xp,
                      int64_t C) {
  TORCH_CHECK(g.is_cuda() && g.dim() == 4 &&
                  g.scalar_type() == at::kBFloat16 &&
                  g.is_contiguous(at::MemoryFormat::ChannelsLast),
              "conv_wgrad: bf16 channels_last grad");
  int N = (int)g.size(0), K = (int)g.size(1);
  int H = (int)g.size(2), W = (int)g.size(3);
  TORCH_CHECK(xp.size(0) == N && xp.size(1) == H && xp.size(2) == W,
              "conv_wgrad: packed activations shape");
  TORCH_CHECK(((int64_t)N * H * W) % 16 == 0 && K % 32 == 0 && C % 32 == 0,
              "conv_wgrad: shape constraints");
  int CW = (int)xp.size(3);
  auto dw = at::zeros({K, C, 3, 3}, g.options().dtype(at::kFloat));
  bdbnn_conv_wgrad(g.data_ptr(), (const uint32_t*)xp.data_ptr<int>(),
                   dw.data_ptr<float>(), N, H, W, (int)C, K, CW,
                   cur_stream());
  return dw;
}

// ---------------- kurtosis ----------------

std::vector<at::Tensor> kurtosis_fwd(const std::vector<at::Tensor>& ws,
                                     const at::Tensor& targets) {
  auto meta = make_meta(ws);
  auto dev = ws[0].device();
  auto sched = build_schedule(ws, 32 * 1024, dev);
  int L = (int)ws.size();
  auto opts = ws[0].options();
  auto mom = at::zeros({L, 4}, opts.dtype(at::kDouble));
  auto stats = at::empty({L, 4}, opts);
  auto losses = at::empty({L}, opts);
  auto kurts = at::empty({L}, opts);
  auto tgt = targets.to(dev, at::kFloat).contiguous();
  bdbnn_kurtosis_fwd(&meta, sched.bt.data_ptr<int>(),
                     sched.bo.data_ptr<int64_t>(), sched.n_blocks,
                     mom.data_ptr<double>(), tgt.data_ptr<float>(),
                     stats.data_ptr<float>(), losses.data_ptr<float>(),
                     kurts.data_ptr<float>(), cur_stream());
  return {losses, kurts, stats};
}

std::vector<at::Tensor> kurtosis_bwd(const std::vector<at::Tensor>& ws,
                                     const at::Tensor& stats,
                                     const at::Tensor& targets,
                                     const at::Tensor& gscale) {
  auto meta = make_meta(ws);
  auto dev = ws[0].device();
  auto sched = build_schedule(ws, 32 * 1024, dev);
  std::vector<at::Tensor> grads;
  for (auto& w : ws)  // preserve_format: grad layout == weight layout
    grads.push_back(at::empty_like(w, w.options(),
                                   at::MemoryFormat::Preserve));
  auto gp = make_ptrs(grads);
  auto tgt = targets.to(dev, at::kFloat).contiguous();
  auto st = stats.contiguous();
  // gscale is a 0-dim DEVICE tensor: no host sync in the backward
  auto gs = gscale.to(dev, at::kFloat).contiguous();
  bdbnn_kurtosis_bwd(&meta, &gp, sched.bt.data_ptr<int>(),
                     sched.bo.data_ptr<int64_t>(), sched.n_blocks,
                     st.data_ptr<float>(), tgt.data_ptr<float>(),
                     gs.data_ptr<float>(), cur_stream());
  return grads;
}

// ---------------- weight KD ----------------

at::Tensor weight_kd_fwd(const std::vector<at::Tensor>& ws,
                         const std::vector<at::Tensor>& wt) {
  auto meta = make_meta(ws);
  auto sched = build_schedule(ws, 32 * 1024, ws[0].device());
  auto s_ptr = make_ptrs(ws);
  std::vector<at::Tensor> wt_c;
  for (auto& t : wt) wt_c.push_back(t.contiguous());
  auto t_ptr = make_ptrs(wt_c);
  auto out = at::zeros({}, ws[0].options().dtype(at::kDouble));
  bdbnn_weight_kd_fwd(&meta, &s_ptr, &t_ptr, sched.bt.data_ptr<int>(),
                      sched.bo.data_ptr<int64_t>(), sched.n_blocks,
                      out.data_ptr<double>(), cur_stream());
  return out.to(at::kFloat);
}

std::vector<at::Tensor> weight_kd_bwd(const std::vector<at::Tensor>& wt,
                                      const at::Tensor& gscale) {
  auto meta = make_meta(wt);
  auto sched = build_schedule(wt, 32 * 1024, wt[0].device());
  auto t_ptr = make_ptrs(wt);
  std::vector<at::Tensor> grads;
  for (auto& t : wt)
    grads.push_back(at::empty_like(t, t.options(),
                                   at::MemoryFormat::Preserve));
  auto gp = make_ptrs(grads);
  auto gs = gscale.to(wt[0].device(), at::kFloat).contiguous();
  bdbnn_weight_kd_bwd(&meta, &t_ptr, &gp, sched.bt.data_ptr<int>(),
                      sched.bo.data_ptr<int64_t>(), sched.n_blocks,
                      gs.data_ptr<float>(), cur_stream());
  return grads;
}

// ---------------- fused optimizers ----------------

void fused_sgd(const std::vector<at::Tensor>& p,
               const std::vector<at::Tensor>& g,
               const std::vector<at::Tensor>& buf, double lr, double momentum,
               double wd) {
  auto meta = make_meta(p);
  auto sched = build_schedule(p, 32 * 1024, p[0].device());
  auto pp = make_ptrs(p);
  auto gg = make_ptrs(g);   // python aligns grad layout to the param's
  auto bb = make_ptrs(buf);
  bdbnn_fused_sgd(&meta, &pp, &gg, &bb, sched.bt.data_ptr<int>(),
                  sched.bo.data_ptr<int64_t>(), sched.n_blocks, (float)lr,
                  (float)momentum, (float)wd, cur_stream());
}

void fused_adam(const std::vector<at::Tensor>& p,
                const std::vector<at::Tensor>& g,
                const std::vector<at::Tensor>& m1,
                const std::vector<at::Tensor>& m2, double lr, double beta1,
                double beta2, double eps, double wd, double bc1, double bc2) {
  auto meta = make_meta(p);
  auto sched = build_schedule(p, 32 * 1024, p[0].device());
  auto pp = make_ptrs(p);
  auto gg = make_ptrs(g);   // python aligns grad layout to the param's
  auto mm1 = make_ptrs(m1);
  auto mm2 = make_ptrs(m2);
  bdbnn_fused_adam(&meta, &pp, &gg, &mm1, &mm2, sched.bt.data_ptr<int>(),
                   sched.bo.data_ptr<int64_t>(), sched.n_blocks, (float)lr,
                   (float)beta1, (float)beta2, (float)eps, (float)wd,
                   (float)bc1, (float)bc2, cur_stream());
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("sign_pack_nhwc", &sign_pack_nhwc, "sign+bitpack NHWC activations");
  m.def("weight_pack", &weight_pack,
        "pack conv weights -> (bits, alpha, pad-correction table)");
  m.def("binsign_decode", &binsign_decode, "elementwise +-1 decode");
  m.def("ste_mask_mul", &ste_mask_mul, "quantizer backward mask multiply");
  m.def("sign_mask_pack_nhwc", &sign_mask_pack_nhwc,
        "pack sign + clip-STE mask bitplanes in one pass");
  m.def("decode_packed", &decode_packed, "packed signs -> +-1 NHWC tensor");
  m.def("mask_mul_packed", &mask_mul_packed, "dx = mask_bit ? g : 0");
  m.def("weight_decode", &weight_decode, "packed weights -> alpha*(+-1)");
  m.def("xnor_conv_fwd", &xnor_conv_fwd, "bit-packed XNOR+popcount conv");
  m.def("stem_conv_fwd", &stem_conv_fwd,
        "MFMA 7x7/2 stem conv fwd (returns out, packed x4)");
  m.def("stem_conv_wrw", &stem_conv_wrw, "MFMA stem conv weight grad");
  m.def("kd_logit_fwd", &kd_logit_fwd, "fused logit-KD fwd");
  m.def("kd_logit_bwd", &kd_logit_bwd, "fused logit-KD bwd");
  m.def("ce_fwd", &ce_fwd, "fused cross-entropy fwd");
  m.def("ce_bwd", &ce_bwd, "fused cross-entropy bwd");
  m.def("maxpool_fwd", &maxpool_fwd, "fused NHWC maxpool fwd (+u8 idx)");
  m.def("maxpool_bwd", &maxpool_bwd, "gather-based NHWC maxpool bwd");
  m.def("prelu_fwd", &prelu_fwd, "fused NHWC per-channel PReLU fwd");
  m.def("prelu_bwd", &prelu_bwd, "fused NHWC per-channel PReLU bwd");
  m.def("bn_act_fwd_train", &bn_act_fwd_train,
        "fused BN(+add)(+act) training forward");
  m.def("bn_act_bwd", &bn_act_bwd, "fused BN(+add)(+act) backward");
  m.def("bn_act_eval", &bn_act_eval, "fused BN(+add)(+act) eval forward");
  m.def("conv_dgrad2", &conv_dgrad2,
        "MFMA bf16 dgrad v2: halo-staged 9-tap implicit GEMM with fused "
        "clip-STE mask (3x3/s1/p1); optional acc adds a same-shape "
        "skip-gradient tensor in the epilogue",
        py::arg("g"), py::arg("wd"), py::arg("mp"), py::arg("C"),
        py::arg("acc") = py::none());
  m.def("dgrad_weight_decode", &dgrad_weight_decode,
        "packed bits -> mirrored transposed +-alpha bf16 [9][C][K]");
  m.def("dgrad2_supported",
        [](int64_t H, int64_t W, int64_t C, int64_t K) {
          return dgrad2_ok((int)H, (int)W, (int)C, (int)K);
        },
        "shape support predicate for conv_dgrad2");
  m.def("conv_wgrad2", &conv_wgrad2,
        "MFMA bf16 wgrad v2: all-9-tap block GEMM from sign BITS "
        "(3x3/s1/p1) -> fp32 [9][C][K]");
  m.def("repack_cplane", &repack_cplane,
        "activation sign bits -> padded per-channel u64 row planes");
  m.def("wgrad_finish", &wgrad_finish,
        "dwT transpose to [K][C][3][3] + |w|<=1 STE mask, one pass");
  m.def("conv_dgrad", &conv_dgrad,
        "EXPERIMENTAL MFMA bf16 dgrad (3x3/s1/p1, packed weights)");
  m.def("conv_wgrad", &conv_wgrad,
        "EXPERIMENTAL MFMA bf16 wgrad (3x3/s1/p1, packed activations)");
  m.def("kurtosis_fwd", &kurtosis_fwd, "fused multi-tensor kurtosis fwd");
  m.def("kurtosis_bwd", &kurtosis_bwd, "fused multi-tensor kurtosis bwd");
  m.def("weight_kd_fwd", &weight_kd_fwd, "fused weight-space KD fwd");
  m.def("weight_kd_bwd", &weight_kd_bwd, "fused weight-space KD bwd");
  m.def("fused_sgd", &fused_sgd, "fused multi-tensor SGD-momentum step");
  m.def("fused_adam", &fused_adam, "fused multi-tensor Adam step");
}
