// Torch bindings for fengshen_amd HIP kernels (gfx950).
// Kernels live in kernels.hip / flash_attn.hip as extern "C" launchers taking
// raw pointers + hipStream_t; this file owns all Tensor plumbing.
#include <torch/extension.h>

#include <hip/hip_runtime.h>
#include <c10/hip/HIPStream.h>

#include <vector>

enum FsDtype { FS_F32 = 0, FS_BF16 = 1, FS_F16 = 2 };

static int fs_dtype(const at::Tensor& t) {
  switch (t.scalar_type()) {
    case at::kFloat: return FS_F32;
    case at::kBFloat16: return FS_BF16;
    case at::kHalf: return FS_F16;
    default: TORCH_CHECK(false, "unsupported dtype ", t.scalar_type());
  }
}

static hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

extern "C" {
void fs_rms_norm_fwd(const void*, const void*, void*, float*, int, int, float,
                     int, hipStream_t);
void fs_rms_norm_bwd(const void*, const void*, const void*, const float*,
                     void*, float*, int, int, int, hipStream_t);
void fs_layer_norm_fwd(const void*, const void*, const void*, void*, float*,
                       float*, int, int, float, int, hipStream_t);
void fs_layer_norm_bwd(const void*, const void*, const void*, const float*,
                       const float*, void*, float*, float*, int, int, int,
                       hipStream_t);
void fs_scaled_softmax_fwd(const void*, const unsigned char*, void*, float,
                           int, int, int, int, int, int, int, int,
                           hipStream_t);
void fs_scaled_softmax_bwd(const void*, const void*, void*, float, int, int,
                           int, hipStream_t);
void fs_rope(const void*, void*, const float*, const float*, long, int, int,
             int, int, int, hipStream_t);
void fs_swiglu_fwd(const void*, void*, long, int, int, hipStream_t);
void fs_swiglu_bwd(const void*, const void*, void*, long, int, int,
                   hipStream_t);
void fs_bias_gelu(const void*, const void*, const void*, void*, long, int, int,
                  int, hipStream_t);
void fs_fused_adamw(float*, const void*, float*, float*, void*, long, float,
                    float, float, float, float, int, int, int, hipStream_t);
void fs_flash_attn_fwd(const void*, const void*, const void*, void*, float*,
                       const int*, int, int, int, int, int, int, float,
                       float, unsigned long long, hipStream_t);
void fs_flash_attn_bwd(const void*, const void*, const void*, const void*,
                       const void*, const float*, void*, void*, void*, float*,
                       const int*, int, int, int, int, int, int, float,
                       float, unsigned long long, hipStream_t);
void fs_bf16_gemv(const void*, const void*, void*, int, int, int,
                  hipStream_t);
void fs_add_rms_norm(const void*, const void*, const void*, void*, void*,
                     int, int, float, hipStream_t);
void fs_decode_attn(const void*, void*, void*, const float*, const float*,
                    const void*, void*, int, int, int, int, float, int,
                    hipStream_t);
void fs_w8_gemv(const void*, const float*, const void*, void*, int, int, int,
                hipStream_t);
void fs_vocab_ce_fwd(const void*, const long*, float*, float*, float*,
                     long, int, int, int, hipStream_t);
void fs_vocab_ce_bwd(const void*, const long*, const float*, const float*,
                     const float*, void*, long, int, int, int, hipStream_t);
void fs_flash_attn_fwd_v3(const void*, const void*, const void*, void*,
                          float*, const int*, int, int, int, int, int, float,
                          float, unsigned long long, hipStream_t);
void fs_flash_attn_bwd_v3(const void*, const void*, const void*, const void*,
                          const void*, const float*, void*, void*, void*,
                          float*, const int*, int, int, int, int, int, float,
                          float, unsigned long long, hipStream_t);
}

// ---------------------------------------------------------------------------
static std::vector<at::Tensor> rms_norm_fwd(at::Tensor x, at::Tensor w,
                                            double eps) {
  TORCH_CHECK(x.is_contiguous() && w.is_contiguous());
  const int H = x.size(-1);
  TORCH_CHECK(H % 8 == 0, "hidden must be divisible by 8");
  const long rows = x.numel() / H;
  auto out = at::empty_like(x);
  auto invrms = at::empty({rows}, x.options().dtype(at::kFloat));
  fs_rms_norm_fwd(x.data_ptr(), w.data_ptr(), out.data_ptr(),
                  invrms.data_ptr<float>(), (int)rows, H, (float)eps,
                  fs_dtype(x), cur_stream());
  return {out, invrms};
}

static std::vector<at::Tensor> rms_norm_bwd(at::Tensor gy, at::Tensor x,
                                            at::Tensor w, at::Tensor invrms) {
  const int H = x.size(-1);
  const long rows = x.numel() / H;
  auto gx = at::empty_like(x);
  auto gw32 = at::zeros({H}, x.options().dtype(at::kFloat));
  fs_rms_norm_bwd(gy.data_ptr(), x.data_ptr(), w.data_ptr(),
                  invrms.data_ptr<float>(), gx.data_ptr(),
                  gw32.data_ptr<float>(), (int)rows, H, fs_dtype(x),
                  cur_stream());
  return {gx, gw32.to(w.scalar_type())};
}

static std::vector<at::Tensor> layer_norm_fwd(at::Tensor x, at::Tensor w,
                                              c10::optional<at::Tensor> b,
                                              double eps) {
  TORCH_CHECK(x.is_contiguous());
  const int H = x.size(-1);
  TORCH_CHECK(H % 8 == 0, "hidden must be divisible by 8");
  const long rows = x.numel() / H;
  auto out = at::empty_like(x);
  auto mean = at::empty({rows}, x.options().dtype(at::kFloat));
  auto invstd = at::empty({rows}, x.options().dtype(at::kFloat));
  fs_layer_norm_fwd(x.data_ptr(), w.data_ptr(),
                    b.has_value() ? b->data_ptr() : nullptr, out.data_ptr(),
                    mean.data_ptr<float>(), invstd.data_ptr<float>(),
                    (int)rows, H, (float)eps, fs_dtype(x), cur_stream());
  return {out, mean, invstd};
}

static std::vector<at::Tensor> layer_norm_bwd(at::Tensor gy, at::Tensor x,
                                              at::Tensor w, at::Tensor mean,
                                              at::Tensor invstd) {
  const int H = x.size(-1);
  const long rows = x.numel() / H;
  auto gx = at::empty_like(x);
  auto gw32 = at::zeros({H}, x.options().dtype(at::kFloat));
  auto gb32 = at::zeros({H}, x.options().dtype(at::kFloat));
  fs_layer_norm_bwd(gy.data_ptr(), x.data_ptr(), w.data_ptr(),
                    mean.data_ptr<float>(), invstd.data_ptr<float>(),
                    gx.data_ptr(), gw32.data_ptr<float>(),
                    gb32.data_ptr<float>(), (int)rows, H, fs_dtype(x),
                    cur_stream());
  return {gx, gw32.to(w.scalar_type()), gb32.to(w.scalar_type())};
}

// x: [b, np, sq, sk]; mask: uint8/bool [mb, 1, sq, sk] (1 = masked)
static at::Tensor scaled_masked_softmax_fwd(at::Tensor x,
                                            c10::optional<at::Tensor> mask,
                                            double scale) {
  TORCH_CHECK(x.dim() == 4 && x.is_contiguous());
  const int sk = x.size(3), sq = x.size(2), np = x.size(1), b = x.size(0);
  auto out = at::empty_like(x);
  const unsigned char* mp = nullptr;
  at::Tensor m8;
  int mb = 1, mode = 0, mask_sq = 1;
  if (mask.has_value()) {
    m8 = mask->to(at::kByte).contiguous();
    TORCH_CHECK(m8.dim() == 4 && m8.size(3) == sk &&
                    (m8.size(2) == sq || m8.size(2) == 1),
                "mask must be [mb,1,sq|1,sk]");
    mb = m8.size(0);
    mask_sq = m8.size(2);
    TORCH_CHECK(mb == 1 || mb == b, "mask batch must be 1 or b");
    mp = m8.data_ptr<unsigned char>();
    mode = 1;
  }
  fs_scaled_softmax_fwd(x.data_ptr(), mp, out.data_ptr(), (float)scale,
                        b * np * sq, sk, sq, np, mb, mask_sq, mode,
                        fs_dtype(x), cur_stream());
  return out;
}

// x: [ab, sq, sk] with sq == sk, causal mask generated in-kernel
static at::Tensor scaled_causal_softmax_fwd(at::Tensor x, double scale) {
  TORCH_CHECK(x.dim() == 3 && x.is_contiguous());
  const int sk = x.size(2), sq = x.size(1), ab = x.size(0);
  TORCH_CHECK(sq == sk, "causal softmax needs sq == sk");
  auto out = at::empty_like(x);
  fs_scaled_softmax_fwd(x.data_ptr(), nullptr, out.data_ptr(), (float)scale,
                        ab * sq, sk, sq, 1, 1, sq, 2, fs_dtype(x),
                        cur_stream());
  return out;
}

static at::Tensor scaled_softmax_bwd(at::Tensor gy, at::Tensor y,
                                     double scale) {
  TORCH_CHECK(gy.is_contiguous() && y.is_contiguous());
  const int sk = y.size(-1);
  const long rows = y.numel() / sk;
  auto gx = at::empty_like(y);
  fs_scaled_softmax_bwd(gy.data_ptr(), y.data_ptr(), gx.data_ptr(),
                        (float)scale, (int)rows, sk, fs_dtype(y),
                        cur_stream());
  return gx;
}

// q,k: [b, np, s, hn]; cos/sin: [S, hn] fp32
static std::vector<at::Tensor> rope_fwd(at::Tensor q, at::Tensor k,
                                        at::Tensor cos, at::Tensor sin,
                                        int64_t offset) {
  TORCH_CHECK(q.is_contiguous() && k.is_contiguous());
  TORCH_CHECK(cos.scalar_type() == at::kFloat && cos.is_contiguous());
  const int hn = q.size(-1), s = q.size(-2);
  TORCH_CHECK((hn / 2) % 8 == 0, "head_dim/2 must be divisible by 8");
  auto qo = at::empty_like(q);
  auto ko = at::empty_like(k);
  const long bnq = q.numel() / ((long)s * hn);
  const long bnk = k.numel() / ((long)s * hn);
  fs_rope(q.data_ptr(), qo.data_ptr(), cos.data_ptr<float>(),
          sin.data_ptr<float>(), bnq, s, hn, (int)offset, 0, fs_dtype(q),
          cur_stream());
  fs_rope(k.data_ptr(), ko.data_ptr(), cos.data_ptr<float>(),
          sin.data_ptr<float>(), bnk, s, hn, (int)offset, 0, fs_dtype(k),
          cur_stream());
  return {qo, ko};
}

static std::vector<at::Tensor> rope_bwd(at::Tensor gq, at::Tensor gk,
                                        at::Tensor cos, at::Tensor sin,
                                        int64_t offset) {
  const int hn = gq.size(-1), s = gq.size(-2);
  auto qo = at::empty_like(gq);
  auto ko = at::empty_like(gk);
  const long bnq = gq.numel() / ((long)s * hn);
  const long bnk = gk.numel() / ((long)s * hn);
  fs_rope(gq.data_ptr(), qo.data_ptr(), cos.data_ptr<float>(),
          sin.data_ptr<float>(), bnq, s, hn, (int)offset, 1, fs_dtype(gq),
          cur_stream());
  fs_rope(gk.data_ptr(), ko.data_ptr(), cos.data_ptr<float>(),
          sin.data_ptr<float>(), bnk, s, hn, (int)offset, 1, fs_dtype(gk),
          cur_stream());
  return {qo, ko};
}

static at::Tensor swiglu_fwd(at::Tensor packed) {
  TORCH_CHECK(packed.is_contiguous());
  const int h2 = packed.size(-1);
  TORCH_CHECK(h2 % 16 == 0, "packed dim must be divisible by 16");
  const int h = h2 / 2;
  const long rows = packed.numel() / h2;
  auto sizes = packed.sizes().vec();
  sizes.back() = h;
  auto out = at::empty(sizes, packed.options());
  fs_swiglu_fwd(packed.data_ptr(), out.data_ptr(), rows, h, fs_dtype(packed),
                cur_stream());
  return out;
}

static at::Tensor swiglu_bwd(at::Tensor gy, at::Tensor packed) {
  const int h2 = packed.size(-1);
  const int h = h2 / 2;
  const long rows = packed.numel() / h2;
  auto gpacked = at::empty_like(packed);
  fs_swiglu_bwd(gy.contiguous().data_ptr(), packed.data_ptr(),
                gpacked.data_ptr(), rows, h, fs_dtype(packed), cur_stream());
  return gpacked;
}

static at::Tensor bias_gelu_fwd(at::Tensor x, at::Tensor bias) {
  TORCH_CHECK(x.is_contiguous() && bias.is_contiguous());
  const int h = x.size(-1);
  TORCH_CHECK(h % 8 == 0);
  const long rows = x.numel() / h;
  auto out = at::empty_like(x);
  fs_bias_gelu(x.data_ptr(), bias.data_ptr(), nullptr, out.data_ptr(), rows, h,
               0, fs_dtype(x), cur_stream());
  return out;
}

static at::Tensor bias_gelu_bwd(at::Tensor gy, at::Tensor x, at::Tensor bias) {
  const int h = x.size(-1);
  const long rows = x.numel() / h;
  auto gx = at::empty_like(x);
  fs_bias_gelu(x.data_ptr(), bias.data_ptr(), gy.data_ptr(), gx.data_ptr(),
               rows, h, 1, fs_dtype(x), cur_stream());
  return gx;
}

static void fused_adamw(at::Tensor master, at::Tensor grad, at::Tensor m,
                        at::Tensor v, at::Tensor out_param, double lr,
                        double beta1, double beta2, double eps, double wd,
                        int64_t step) {
  TORCH_CHECK(master.scalar_type() == at::kFloat);
  TORCH_CHECK(master.is_contiguous() && grad.is_contiguous());
  void* outp = out_param.data_ptr() == master.data_ptr()
                   ? nullptr : out_param.data_ptr();
  fs_fused_adamw(master.data_ptr<float>(), grad.data_ptr(),
                 m.data_ptr<float>(), v.data_ptr<float>(), outp,
                 master.numel(), (float)lr, (float)beta1, (float)beta2,
                 (float)eps, (float)wd, (int)step, fs_dtype(grad),
                 outp ? fs_dtype(out_param) : FS_F32, cur_stream());
}

static const int kFlashDims[] = {40, 64, 80, 96, 128, 160};

static bool flash_dim_ok(int d) {
  for (int x : kFlashDims)
    if (x == d) return true;
  return false;
}

// q: [b, h, sq, d] bf16 contiguous; k,v: [b, h, sk, d].
// causal: sq == sk, sq % 64 == 0, klens must be absent.
// non-causal: cross-attention (sq != sk) and ragged sk allowed; optional
// klens int32 [b] = per-batch key length (suffix padding mask).
static std::vector<at::Tensor> flash_attn_fwd(
    at::Tensor q, at::Tensor k, at::Tensor v, double scale, bool causal,
    c10::optional<at::Tensor> klens, double drop_p, int64_t seed) {
  TORCH_CHECK(q.is_contiguous() && k.is_contiguous() && v.is_contiguous());
  TORCH_CHECK(q.scalar_type() == at::kBFloat16, "flash_attn: bf16 only");
  TORCH_CHECK(q.dim() == 4 && flash_dim_ok(q.size(3)),
              "flash_attn: head_dim must be one of 40/64/80/96/128/160");
  const int b = q.size(0), h = q.size(1), sq = q.size(2), d = q.size(3);
  const int sk = k.size(2);
  if (causal) {
    TORCH_CHECK(sq == sk && sq % 64 == 0 && !klens.has_value(),
                "flash_attn causal: sq==sk, sq%64==0, no klens");
  } else {
    TORCH_CHECK(sq % 16 == 0 && sq >= 16,
                "flash_attn: sq must be a multiple of 16");
  }
  const int* klp = nullptr;
  if (klens.has_value()) {
    TORCH_CHECK(klens->scalar_type() == at::kInt && klens->numel() == b
                && klens->is_contiguous());
    klp = klens->data_ptr<int>();
  }
  auto o = at::empty_like(q);
  auto lse = at::empty({b, h, sq}, q.options().dtype(at::kFloat));
  fs_flash_attn_fwd(q.data_ptr(), k.data_ptr(), v.data_ptr(), o.data_ptr(),
                    lse.data_ptr<float>(), klp, b, h, sq, sk, d,
                    causal ? 1 : 0, (float)scale, (float)drop_p,
                    (unsigned long long)seed, cur_stream());
  return {o, lse};
}

static std::vector<at::Tensor> flash_attn_bwd(
    at::Tensor q, at::Tensor k, at::Tensor v, at::Tensor o, at::Tensor dout,
    at::Tensor lse, double scale, bool causal,
    c10::optional<at::Tensor> klens, double drop_p, int64_t seed) {
  const int b = q.size(0), h = q.size(1), sq = q.size(2), d = q.size(3);
  const int sk = k.size(2);
  const int* klp = nullptr;
  if (klens.has_value()) klp = klens->data_ptr<int>();
  auto dq = at::empty_like(q);
  auto dk = at::empty_like(k);
  auto dv = at::empty_like(v);
  auto delta = at::empty({b, h, sq}, q.options().dtype(at::kFloat));
  fs_flash_attn_bwd(q.data_ptr(), k.data_ptr(), v.data_ptr(), o.data_ptr(),
                    dout.contiguous().data_ptr(), lse.data_ptr<float>(),
                    dq.data_ptr(), dk.data_ptr(), dv.data_ptr(),
                    delta.data_ptr<float>(), klp, b, h, sq, sk, d,
                    causal ? 1 : 0, (float)scale, (float)drop_p,
                    (unsigned long long)seed, cur_stream());
  return {dq, dk, dv};
}

// x [b, in] bf16, q8 [out, in] int8, scale [out] fp32 -> y [b, out] bf16
static at::Tensor w8_gemv(at::Tensor q8, at::Tensor scale, at::Tensor x) {
  TORCH_CHECK(q8.scalar_type() == at::kChar && x.scalar_type() == at::kBFloat16);
  TORCH_CHECK(q8.is_contiguous() && x.is_contiguous());
  const int out = q8.size(0), in = q8.size(1);
  TORCH_CHECK(in % 16 == 0, "in_features must be divisible by 16");
  const int b = x.numel() / in;
  auto y = at::empty({b, out}, x.options());
  fs_w8_gemv(q8.data_ptr(), scale.data_ptr<float>(), x.data_ptr(),
             y.data_ptr(), b, in, out, cur_stream());
  return y;
}

// x [b, in] bf16 (b <= 8), w [out, in] bf16 -> y [b, out] bf16
static at::Tensor bf16_gemv(at::Tensor w, at::Tensor x) {
  TORCH_CHECK(w.scalar_type() == at::kBFloat16 &&
              x.scalar_type() == at::kBFloat16);
  TORCH_CHECK(w.is_contiguous() && x.is_contiguous());
  const int out = w.size(0), in = w.size(1);
  TORCH_CHECK(in % 16 == 0, "in_features must be divisible by 16");
  const int b = x.numel() / in;
  TORCH_CHECK(b >= 1 && b <= 8, "bf16_gemv: batch must be 1..8");
  auto y = at::empty({b, out}, x.options());
  fs_bf16_gemv(w.data_ptr(), x.data_ptr(), y.data_ptr(), b, in, out,
               cur_stream());
  return y;
}

// Fused decode step: qkv [b, 3*H] bf16, kc/vc [b, nh, L, d] bf16 caches
// (written in-place at pos), cos/sin [max_len, d] fp32, pos [1] long.
// Returns ctx [b, H] bf16.
static at::Tensor decode_attn(at::Tensor qkv, at::Tensor kc, at::Tensor vc,
                              at::Tensor cos_t, at::Tensor sin_t,
                              at::Tensor pos, double scale, bool rope) {
  TORCH_CHECK(qkv.scalar_type() == at::kBFloat16 && qkv.is_contiguous());
  TORCH_CHECK(kc.is_contiguous() && vc.is_contiguous());
  TORCH_CHECK(cos_t.scalar_type() == at::kFloat &&
              sin_t.scalar_type() == at::kFloat);
  TORCH_CHECK(pos.scalar_type() == at::kLong);
  const int b = kc.size(0), nh = kc.size(1), L = kc.size(2), d = kc.size(3);
  TORCH_CHECK(d == 64 || d == 128, "decode_attn: head dim must be 64/128");
  TORCH_CHECK(qkv.numel() == (long)b * 3 * nh * d, "qkv shape mismatch");
  auto ctx = at::empty({b, nh * d}, qkv.options());
  fs_decode_attn(qkv.data_ptr(), kc.data_ptr(), vc.data_ptr(),
                 cos_t.data_ptr<float>(), sin_t.data_ptr<float>(),
                 pos.data_ptr(), ctx.data_ptr(), b, nh, d, L, (float)scale,
                 rope ? 1 : 0, cur_stream());
  return ctx;
}

// a,b [rows, H] bf16 -> (sum=a+b, y=rmsnorm(sum)*w), inference-only
static std::vector<at::Tensor> add_rms_norm(at::Tensor a, at::Tensor b,
                                            at::Tensor w, double eps) {
  TORCH_CHECK(a.scalar_type() == at::kBFloat16 && a.is_contiguous() &&
              b.is_contiguous() && w.is_contiguous());
  const int H = a.size(-1);
  TORCH_CHECK(H % 8 == 0 && w.numel() == H && b.sizes() == a.sizes());
  const int rows = a.numel() / H;
  auto sum_out = at::empty_like(a);
  auto y = at::empty_like(a);
  fs_add_rms_norm(a.data_ptr(), b.data_ptr(), w.data_ptr(),
                  sum_out.data_ptr(), y.data_ptr(), rows, H, (float)eps,
                  cur_stream());
  return {sum_out, y};
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, mod) {
  mod.def("w8_gemv", &w8_gemv);
  mod.def("bf16_gemv", &bf16_gemv);
  mod.def("decode_attn", &decode_attn);
  mod.def("add_rms_norm", &add_rms_norm);
  mod.def("flash_attn_fwd", &flash_attn_fwd);
  mod.def("vocab_ce_fwd", [](at::Tensor logits2d, at::Tensor targets,
                             int64_t vstart, int64_t vend) {
    TORCH_CHECK(logits2d.is_contiguous() && logits2d.dim() == 2
                && logits2d.scalar_type() == at::kBFloat16);
    TORCH_CHECK(targets.scalar_type() == at::kLong
                && targets.is_contiguous());
    const long n = logits2d.size(0);
    const int w = logits2d.size(1);
    TORCH_CHECK(w % 8 == 0, "vocab shard must be divisible by 8");
    auto opts = logits2d.options().dtype(at::kFloat);
    auto m = at::empty({n}, opts);
    auto z = at::empty({n}, opts);
    auto pred = at::empty({n}, opts);
    fs_vocab_ce_fwd(logits2d.data_ptr(), targets.data_ptr<long>(),
                    m.data_ptr<float>(), z.data_ptr<float>(),
                    pred.data_ptr<float>(), n, w, (int)vstart, (int)vend,
                    cur_stream());
    return std::vector<at::Tensor>{m, z, pred};
  });
  mod.def("vocab_ce_bwd", [](at::Tensor logits2d, at::Tensor targets,
                             at::Tensor m, at::Tensor z, at::Tensor gout,
                             int64_t vstart, int64_t vend) {
    const long n = logits2d.size(0);
    const int w = logits2d.size(1);
    auto dl = at::empty_like(logits2d);
    fs_vocab_ce_bwd(logits2d.data_ptr(), targets.data_ptr<long>(),
                    m.data_ptr<float>(), z.data_ptr<float>(),
                    gout.contiguous().data_ptr<float>(), dl.data_ptr(), n, w,
                    (int)vstart, (int)vend, cur_stream());
    return dl;
  });
  mod.def("flash_attn_bwd_v3", [](at::Tensor q, at::Tensor k, at::Tensor v,
                                  at::Tensor o, at::Tensor dout,
                                  at::Tensor lse, double scale, bool causal,
                                  c10::optional<at::Tensor> klens,
                                  double drop_p, int64_t seed) {
    const int b = q.size(0), h = q.size(1), s = q.size(2), d = q.size(3);
    TORCH_CHECK((d == 128 || d == 96 || d == 64) && s % 64 == 0);
    const int* klp = nullptr;
    if (klens.has_value()) klp = klens->data_ptr<int>();
    auto dq = at::empty_like(q);
    auto dk = at::empty_like(k);
    auto dv = at::empty_like(v);
    auto delta = at::empty({b, h, s}, q.options().dtype(at::kFloat));
    fs_flash_attn_bwd_v3(q.data_ptr(), k.data_ptr(), v.data_ptr(),
                         o.data_ptr(), dout.contiguous().data_ptr(),
                         lse.data_ptr<float>(), dq.data_ptr(), dk.data_ptr(),
                         dv.data_ptr(), delta.data_ptr<float>(), klp, b, h,
                         s, d, causal ? 1 : 0, (float)scale, (float)drop_p,
                         (unsigned long long)seed, cur_stream());
    return std::vector<at::Tensor>{dq, dk, dv};
  });
  mod.def("flash_attn_fwd_v3", [](at::Tensor q, at::Tensor k, at::Tensor v,
                                  double scale, bool causal,
                                  c10::optional<at::Tensor> klens,
                                  double drop_p, int64_t seed) {
    TORCH_CHECK(q.is_contiguous() && k.is_contiguous() && v.is_contiguous());
    TORCH_CHECK(q.scalar_type() == at::kBFloat16);
    const int b = q.size(0), h = q.size(1), s = q.size(2), d = q.size(3);
    TORCH_CHECK((d == 128 || d == 96 || d == 64) && s % 64 == 0 && s >= 64);
    TORCH_CHECK(k.size(2) == s, "v3 is self-attention (sq == sk)");
    const int* klp = nullptr;
    if (klens.has_value()) {
      TORCH_CHECK(klens->scalar_type() == at::kInt
                  && klens->numel() == b && klens->is_contiguous());
      klp = klens->data_ptr<int>();
    }
    auto o = at::empty_like(q);
    auto lse = at::empty({b, h, s}, q.options().dtype(at::kFloat));
    fs_flash_attn_fwd_v3(q.data_ptr(), k.data_ptr(), v.data_ptr(),
                         o.data_ptr(), lse.data_ptr<float>(), klp, b, h, s,
                         d, causal ? 1 : 0, (float)scale, (float)drop_p,
                         (unsigned long long)seed, cur_stream());
    return std::vector<at::Tensor>{o, lse};
  });
  mod.def("flash_attn_bwd", &flash_attn_bwd);
  mod.def("rms_norm_fwd", &rms_norm_fwd);
  mod.def("rms_norm_bwd", &rms_norm_bwd);
  mod.def("layer_norm_fwd", &layer_norm_fwd);
  mod.def("layer_norm_bwd", &layer_norm_bwd);
  mod.def("scaled_masked_softmax_fwd", &scaled_masked_softmax_fwd);
  mod.def("scaled_causal_softmax_fwd", &scaled_causal_softmax_fwd);
  mod.def("scaled_softmax_bwd", &scaled_softmax_bwd);
  mod.def("rope_fwd", &rope_fwd);
  mod.def("rope_bwd", &rope_bwd);
  mod.def("swiglu_fwd", &swiglu_fwd);
  mod.def("swiglu_bwd", &swiglu_bwd);
  mod.def("bias_gelu_fwd", &bias_gelu_fwd);
  mod.def("bias_gelu_bwd", &bias_gelu_bwd);
  mod.def("fused_adamw", &fused_adamw);
}
