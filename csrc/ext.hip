// Torch bindings for the gansformer_amd gfx950 HIP kernels.
// Replaces the reference's import-time nvcc JIT + tf.load_op_library
// (ref src/dnnlib/tflib/custom_ops.py [R]) with an AOT-built extension.
#include <torch/extension.h>

#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

#include "common.h"

namespace gfa {

struct UfdParams {
  int B, C, H, W;
  int OH, OW;
  int fh, fw;
  int upx, upy, downx, downy;
  int px0, px1, py0, py1;
  float gain;
};

struct ConvParams {
  int B, I, H, W;
  int O, OH, OW;
  int kh, kw, stride, pad;
  int per_sample;
};

struct AttnParams {
  int B, Nq, Nk, D, E;
  float scale;
};

template <typename T>
void launch_fba(T*, const T*, const float*, const T*, long, long, int, int,
                int, float, float, float, hipStream_t);
void launch_fba_mod(bf16*, const bf16*, const float*, const bf16*,
                    const float*, const float*, long, long, int, int, float,
                    float, float, hipStream_t);
template <typename T>
void launch_upfirdn2d(T*, const T*, const float*, const UfdParams&,
                      hipStream_t);
void launch_conv2d_fwd_bf16(bf16*, const bf16*, const bf16*,
                            const ConvParams&, hipStream_t);
bool conv2d_slab_eligible(int I, int O, int H, int W, int OH, int OW,
                          int kh, int kw, int stride, int pad,
                          int per_sample);
void launch_conv2d_fwd_slab_bf16(bf16*, const bf16*, const bf16*, int B,
                                 int I, int H, int W, int O, hipStream_t);
void launch_upfirdn2d_sep4_bf16(bf16*, const bf16*, const float*, int, int,
                                int, int, int, int, int, int, int, int,
                                float, hipStream_t);
bool conv2d_up2_eligible(int I, int O, int H, int W, int kh, int kw,
                         int per_sample);
bool conv2d_s2_eligible(int I, int O, int H, int W, int OH, int OW, int kh,
                        int kw, int stride, int pad, int per_sample);
void launch_conv2d_s2_slab_bf16(bf16*, const bf16*, const bf16*, int B,
                                int I, int H, int W, int O, hipStream_t);
void launch_conv2d_up2_slab_bf16(bf16*, const bf16*, const bf16*, int B,
                                 int I, int H, int W, int O, hipStream_t);
bool conv2d_wgrad_slab_eligible(int I, int O, int H, int W, int OH, int OW,
                                int kh, int kw, int stride, int pad,
                                int per_sample);
int conv2d_wgrad_slab_nsplit(int B, int I, int H, int W, int O, int stride);
void launch_conv2d_wgrad_slab_bf16(float*, const bf16*, const bf16*, int B,
                                   int I, int H, int W, int O, int nsplit,
                                   int stride, hipStream_t);
void launch_conv2d_fwd_f32(float*, const float*, const float*,
                           const ConvParams&, hipStream_t);
int conv2d_wgrad_nsplit(const ConvParams&);
template <typename T>
void launch_conv2d_wgrad(T*, float*, const T*, const T*, const ConvParams&,
                         int, hipStream_t);
void launch_modnorm_fwd_bf16(bf16*, float*, float*, const bf16*,
                             const bf16*, const bf16*, long, int, float,
                             hipStream_t);
void launch_modnorm_bwd_bf16(bf16*, bf16*, const bf16*, const bf16*,
                             const bf16*, long, int, float, hipStream_t);
template <typename T>
void launch_mbstd(float*, const T*, int, int, int, int, int, float,
                  hipStream_t);
template <typename T>
void launch_attn_smalln(T*, const T*, const T*, const T*, const AttnParams&,
                        hipStream_t);
template <typename T>
void launch_attn_longn(T*, const T*, const T*, const T*, float*, float*,
                       float*, float*, const AttnParams&, int, hipStream_t);
struct AttnBwdParams {
  int B, Nq, Nk, D, E;
  float scale;
};
template <typename T>
void launch_attn_smalln_bwd(T*, float*, float*, const T*, const T*, const T*,
                            const T*, const float*, const AttnBwdParams&,
                            hipStream_t);
template <typename T>
void launch_attn_longn_bwd(float*, T*, T*, const T*, const T*, const T*,
                           const T*, const float*, const float*,
                           const AttnBwdParams&, hipStream_t);
struct GemmParams {
  long M;
  int N, K;
};
template <typename T>
void launch_gemm_skinny(T*, const T*, const T*, const GemmParams&, bool,
                        hipStream_t);
template <typename T>
void launch_pack_w_blocked(T*, const T*, int, int, int, hipStream_t);
template <typename T>
void launch_pack_w_o9i(T*, const T*, int, int, int, hipStream_t);

}  // namespace gfa

namespace {

using torch::Tensor;

hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

void check_dtype(const Tensor& t, const char* name) {
  TORCH_CHECK(t.scalar_type() == torch::kFloat32 ||
                  t.scalar_type() == torch::kBFloat16,
              name, ": expected float32 or bfloat16, got ", t.scalar_type());
}

#define DISPATCH_FT(t, NAME, ...)                                         \
  do {                                                                    \
    if ((t).scalar_type() == torch::kFloat32) {                           \
      using scalar_t = float;                                             \
      __VA_ARGS__;                                                        \
    } else {                                                              \
      using scalar_t = gfa::bf16;                                         \
      __VA_ARGS__;                                                        \
    }                                                                     \
  } while (0)

template <typename T>
T* ptr(Tensor& t) { return reinterpret_cast<T*>(t.data_ptr()); }
template <typename T>
const T* cptr(const Tensor& t) { return reinterpret_cast<const T*>(t.data_ptr()); }

Tensor fba(Tensor x, Tensor b, Tensor ref, int64_t act, int64_t grad,
           double alpha, double gain, double clamp) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous());
  check_dtype(x, "fba.x");
  auto out = torch::empty_like(x);
  long n = x.numel();
  long inner = 1;
  int C = 1;
  Tensor b32;
  const float* bptr = nullptr;
  if (grad == 0 && b.numel() > 0) {
    TORCH_CHECK(x.dim() >= 2, "fba: bias needs a channel dim");
    C = (int)x.size(1);
    for (int d = 2; d < x.dim(); ++d) inner *= x.size(d);
    b32 = b.to(torch::kFloat32).contiguous();
    TORCH_CHECK(b32.numel() == C, "fba: bias size mismatch");
    bptr = b32.data_ptr<float>();
  }
  if (grad == 1) {
    TORCH_CHECK(ref.sizes() == x.sizes() && ref.scalar_type() == x.scalar_type(),
                "fba: ref must match x");
  }
  DISPATCH_FT(x, "fba", {
    gfa::launch_fba<scalar_t>(
        ptr<scalar_t>(out), cptr<scalar_t>(x), bptr,
        grad == 1 ? cptr<scalar_t>(ref) : nullptr, n, inner, C, (int)act,
        (int)grad, (float)alpha, (float)gain, (float)clamp, cur_stream());
  });
  return out;
}

Tensor fba_mod(Tensor x, Tensor d, Tensor noise, Tensor bias,
               Tensor sigma, int64_t act, double alpha, double gain,
               double clamp) {
  // y = clamp(act(x*d[b,c] + noise*sigma + bias[c]) * gain); bf16 NCHW
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 4);
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16, "fba_mod: bf16 only");
  const long B = x.size(0), C = x.size(1);
  const long hw = x.size(2) * x.size(3);
  TORCH_CHECK(hw % 8 == 0, "fba_mod: H*W must be 8-aligned");
  TORCH_CHECK(d.is_contiguous() && d.numel() == B * C &&
              d.scalar_type() == torch::kFloat32, "fba_mod: d must be f32 [B,C]");
  const bool has_noise = noise.numel() > 0;
  Tensor sig32;
  if (has_noise) {
    TORCH_CHECK(noise.is_contiguous() && noise.numel() == B * hw &&
                noise.scalar_type() == torch::kBFloat16,
                "fba_mod: noise must be bf16 [B,1,H,W]");
    // sigma stays on device (an .item() here would sync the stream
    // once per synthesis layer)
    sig32 = sigma.to(torch::kFloat32).contiguous();
    TORCH_CHECK(sig32.is_cuda() && sig32.numel() >= 1);
  }
  const bool has_bias = bias.numel() > 0;
  Tensor b32;
  if (has_bias) {
    b32 = bias.to(torch::kFloat32).contiguous();
    TORCH_CHECK(b32.numel() == C, "fba_mod: bias size mismatch");
  }
  auto out = torch::empty_like(x);
  gfa::launch_fba_mod(
      ptr<gfa::bf16>(out), cptr<gfa::bf16>(x), d.data_ptr<float>(),
      has_noise ? cptr<gfa::bf16>(noise) : nullptr,
      has_bias ? b32.data_ptr<float>() : nullptr,
      has_noise ? sig32.data_ptr<float>() : nullptr,
      x.numel(), hw, (int)C, (int)act, (float)alpha, (float)gain,
      (float)clamp, cur_stream());
  return out;
}

Tensor upfirdn2d(Tensor x, Tensor f, int64_t upx, int64_t upy, int64_t downx,
                 int64_t downy, int64_t px0, int64_t px1, int64_t py0,
                 int64_t py1, double gain) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 4);
  TORCH_CHECK(f.is_cuda() && f.is_contiguous() && f.dim() == 2 &&
              f.scalar_type() == torch::kFloat32);
  check_dtype(x, "upfirdn2d.x");
  gfa::UfdParams p;
  p.B = (int)x.size(0); p.C = (int)x.size(1);
  p.H = (int)x.size(2); p.W = (int)x.size(3);
  p.fh = (int)f.size(0); p.fw = (int)f.size(1);
  p.upx = (int)upx; p.upy = (int)upy;
  p.downx = (int)downx; p.downy = (int)downy;
  p.px0 = (int)px0; p.px1 = (int)px1; p.py0 = (int)py0; p.py1 = (int)py1;
  p.gain = (float)gain;
  p.OH = (int)((p.H * p.upy + p.py0 + p.py1 - p.fh) / p.downy + 1);
  p.OW = (int)((p.W * p.upx + p.px0 + p.px1 - p.fw) / p.downx + 1);
  TORCH_CHECK(p.OH > 0 && p.OW > 0, "upfirdn2d: empty output");
  auto out = torch::empty({p.B, p.C, p.OH, p.OW}, x.options());
  DISPATCH_FT(x, "upfirdn2d", {
    gfa::launch_upfirdn2d<scalar_t>(ptr<scalar_t>(out), cptr<scalar_t>(x),
                                    f.data_ptr<float>(), p, cur_stream());
  });
  return out;
}

Tensor upfirdn2d_sep(Tensor x, Tensor f8, int64_t u, int64_t d, int64_t px0,
                     int64_t px1, int64_t py0, int64_t py1, double gain) {
  // separable 4-tap path: f8 = concat(fy[4], fx[4]); u/d symmetric,
  // (u,d) in {(1,1),(2,1),(1,2)}; bf16 only
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 4);
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16, "upfirdn2d_sep: bf16");
  TORCH_CHECK(f8.is_cuda() && f8.is_contiguous() && f8.numel() == 8 &&
              f8.scalar_type() == torch::kFloat32);
  TORCH_CHECK((u == 1 && d == 1) || (u == 2 && d == 1) || (u == 1 && d == 2),
              "upfirdn2d_sep: unsupported up/down");
  const int B = (int)x.size(0), C = (int)x.size(1);
  const int H = (int)x.size(2), W = (int)x.size(3);
  const int OH = (int)((H * u + py0 + py1 - 4) / d + 1);
  const int OW = (int)((W * u + px0 + px1 - 4) / d + 1);
  TORCH_CHECK(OH > 0 && OW > 0, "upfirdn2d_sep: empty output");
  auto out = torch::empty({B, C, OH, OW}, x.options());
  gfa::launch_upfirdn2d_sep4_bf16(
      ptr<gfa::bf16>(out), cptr<gfa::bf16>(x), f8.data_ptr<float>(), B, C, H,
      W, OH, OW, (int)u, (int)d, (int)px0, (int)py0, (float)gain,
      cur_stream());
  return out;
}

gfa::ConvParams conv_params(const Tensor& x, int O, int kh, int kw,
                            int stride, int pad, bool per_sample) {
  gfa::ConvParams p;
  p.B = (int)x.size(0); p.I = (int)x.size(1);
  p.H = (int)x.size(2); p.W = (int)x.size(3);
  p.O = O; p.kh = kh; p.kw = kw; p.stride = stride; p.pad = pad;
  p.per_sample = per_sample ? 1 : 0;
  p.OH = (p.H + 2 * pad - kh) / stride + 1;
  p.OW = (p.W + 2 * pad - kw) / stride + 1;
  TORCH_CHECK(p.OH > 0 && p.OW > 0, "conv2d: empty output");
  return p;
}

Tensor conv2d_fwd(Tensor x, Tensor w, int64_t stride, int64_t pad) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 4);
  TORCH_CHECK(w.is_cuda() && w.is_contiguous());
  TORCH_CHECK(w.dim() == 4 || w.dim() == 5, "conv2d: w must be 4D or 5D");
  TORCH_CHECK(x.scalar_type() == w.scalar_type(), "conv2d: dtype mismatch");
  check_dtype(x, "conv2d.x");
  bool per_sample = (w.dim() == 5);
  if (per_sample) TORCH_CHECK(w.size(0) == x.size(0), "conv2d: batch mismatch");
  int O = (int)w.size(per_sample ? 1 : 0);
  int I = (int)w.size(per_sample ? 2 : 1);
  int kh = (int)w.size(per_sample ? 3 : 2);
  int kw = (int)w.size(per_sample ? 4 : 3);
  TORCH_CHECK(I == x.size(1), "conv2d: channel mismatch");
  auto p = conv_params(x, O, kh, kw, (int)stride, (int)pad, per_sample);
  auto out = torch::empty({p.B, O, p.OH, p.OW}, x.options());
  if (x.scalar_type() == torch::kFloat32) {
    gfa::launch_conv2d_fwd_f32(ptr<float>(out), cptr<float>(x),
                               cptr<float>(w), p, cur_stream());
  } else if (gfa::conv2d_slab_eligible(p.I, p.O, p.H, p.W, p.OH, p.OW, p.kh,
                                       p.kw, p.stride, p.pad, p.per_sample)) {
    // blocked weight layout [mT][cB][9][128][32] (o zero-padded to 128):
    // one wave's A fragment set per (tap, cb) is a contiguous 1 KB read;
    // packed by one kernel launch (pack.hip), not a torch op chain
    int mT = (O + 127) / 128, cB = I / 32;
    auto wb = torch::empty({mT, cB, 9, 128, 32}, w.options());
    gfa::launch_pack_w_blocked<gfa::bf16>(ptr<gfa::bf16>(wb),
                                          cptr<gfa::bf16>(w), O, I, mT,
                                          cur_stream());
    gfa::launch_conv2d_fwd_slab_bf16(ptr<gfa::bf16>(out), cptr<gfa::bf16>(x),
                                     cptr<gfa::bf16>(wb), p.B, p.I, p.H, p.W,
                                     p.O, cur_stream());
  } else if (gfa::conv2d_s2_eligible(p.I, p.O, p.H, p.W, p.OH, p.OW, p.kh,
                                     p.kw, p.stride, p.pad, p.per_sample)) {
    auto wr = torch::empty({O, kh * kw, I}, w.options());
    gfa::launch_pack_w_o9i<gfa::bf16>(ptr<gfa::bf16>(wr), cptr<gfa::bf16>(w),
                                      O, I, kh * kw, cur_stream());
    gfa::launch_conv2d_s2_slab_bf16(ptr<gfa::bf16>(out), cptr<gfa::bf16>(x),
                                    cptr<gfa::bf16>(wr), p.B, p.I, p.H, p.W,
                                    p.O, cur_stream());
  } else {
    gfa::launch_conv2d_fwd_bf16(ptr<gfa::bf16>(out), cptr<gfa::bf16>(x),
                                cptr<gfa::bf16>(w), p, cur_stream());
  }
  return out;
}

Tensor conv2d_up2(Tensor x, Tensor w) {
  // y = conv2d(zero_stuff2(x), w, pad=1): parity-decomposed transposed
  // conv at input resolution (1/4 the MACs of conv at output res)
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 4);
  TORCH_CHECK(w.is_cuda() && w.is_contiguous() && w.dim() == 4);
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16 &&
              w.scalar_type() == torch::kBFloat16,
              "conv2d_up2: bf16 only");
  const int B = (int)x.size(0), I = (int)x.size(1);
  const int H = (int)x.size(2), W = (int)x.size(3);
  const int O = (int)w.size(0);
  TORCH_CHECK(w.size(1) == I && w.size(2) == 3 && w.size(3) == 3);
  TORCH_CHECK(gfa::conv2d_up2_eligible(I, O, H, W, 3, 3, 0),
              "conv2d_up2: shape not eligible (I%32, H%8, W%16)");
  auto wr = torch::empty({O, 9, I}, w.options());
  gfa::launch_pack_w_o9i<gfa::bf16>(ptr<gfa::bf16>(wr), cptr<gfa::bf16>(w),
                                    O, I, 9, cur_stream());
  auto out = torch::empty({B, O, 2 * H, 2 * W}, x.options());
  gfa::launch_conv2d_up2_slab_bf16(ptr<gfa::bf16>(out), cptr<gfa::bf16>(x),
                                   cptr<gfa::bf16>(wr), B, I, H, W, O,
                                   cur_stream());
  return out;
}

Tensor conv2d_wgrad(Tensor x, Tensor dy, int64_t stride, int64_t pad,
                    int64_t kh, int64_t kw, bool per_sample) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 4);
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous() && dy.dim() == 4);
  TORCH_CHECK(x.scalar_type() == dy.scalar_type());
  check_dtype(x, "conv2d_wgrad.x");
  int O = (int)dy.size(1);
  auto p = conv_params(x, O, (int)kh, (int)kw, (int)stride, (int)pad,
                       per_sample);
  TORCH_CHECK(p.OH == dy.size(2) && p.OW == dy.size(3),
              "conv2d_wgrad: dy shape mismatch");
  if (x.scalar_type() == torch::kBFloat16 &&
      gfa::conv2d_wgrad_slab_eligible(p.I, p.O, p.H, p.W, p.OH, p.OW, p.kh,
                                      p.kw, p.stride, p.pad, p.per_sample)) {
    // tap-major kernel accumulates f32 into [O][9][I]; rearrange after
    auto ws = torch::zeros({O, 9, x.size(1)},
                           x.options().dtype(torch::kFloat32));
    int ns = gfa::conv2d_wgrad_slab_nsplit(p.B, p.I, p.H, p.W, p.O,
                                           p.stride);
    gfa::launch_conv2d_wgrad_slab_bf16(ws.data_ptr<float>(),
                                       cptr<gfa::bf16>(x),
                                       cptr<gfa::bf16>(dy), p.B, p.I, p.H,
                                       p.W, p.O, ns, p.stride, cur_stream());
    return ws.reshape({O, 3, 3, x.size(1)})
        .permute({0, 3, 1, 2})
        .contiguous()
        .to(x.scalar_type());
  }
  auto shape = per_sample
                   ? std::vector<int64_t>{x.size(0), O, x.size(1), kh, kw}
                   : std::vector<int64_t>{O, x.size(1), kh, kw};
  int nsplit = gfa::conv2d_wgrad_nsplit(p);
  if (nsplit == 1) {
    auto dw = torch::empty(shape, x.options());
    DISPATCH_FT(x, "conv2d_wgrad", {
      gfa::launch_conv2d_wgrad<scalar_t>(ptr<scalar_t>(dw), nullptr,
                                         cptr<scalar_t>(x), cptr<scalar_t>(dy),
                                         p, 1, cur_stream());
    });
    return dw;
  }
  auto ws = torch::zeros(shape, x.options().dtype(torch::kFloat32));
  DISPATCH_FT(x, "conv2d_wgrad", {
    gfa::launch_conv2d_wgrad<scalar_t>(nullptr, ws.data_ptr<float>(),
                                       cptr<scalar_t>(x), cptr<scalar_t>(dy),
                                       p, nsplit, cur_stream());
  });
  return ws.to(x.scalar_type());
}

std::vector<Tensor> modnorm(Tensor x, Tensor gamma, Tensor beta,
                            double eps) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous());
  TORCH_CHECK(gamma.is_contiguous() && beta.is_contiguous());
  TORCH_CHECK(x.sizes() == gamma.sizes() && x.sizes() == beta.sizes());
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16 &&
              gamma.scalar_type() == torch::kBFloat16 &&
              beta.scalar_type() == torch::kBFloat16,
              "modnorm: bf16 only");
  const long N = x.size(-1);
  TORCH_CHECK(N % 8 == 0 && N > 0, "modnorm: last dim must be 8-aligned");
  const long BC = x.numel() / N;
  auto y = torch::empty_like(x);
  auto mean = torch::empty({BC}, x.options().dtype(torch::kFloat32));
  auto rstd = torch::empty({BC}, x.options().dtype(torch::kFloat32));
  gfa::launch_modnorm_fwd_bf16(ptr<gfa::bf16>(y), mean.data_ptr<float>(),
                               rstd.data_ptr<float>(), cptr<gfa::bf16>(x),
                               cptr<gfa::bf16>(gamma), cptr<gfa::bf16>(beta),
                               BC, (int)N, (float)eps, cur_stream());
  return {y, mean, rstd};
}

std::vector<Tensor> modnorm_bwd(Tensor x, Tensor gamma, Tensor dy,
                                double eps) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && gamma.is_contiguous() &&
              dy.is_contiguous());
  TORCH_CHECK(x.sizes() == gamma.sizes() && x.sizes() == dy.sizes());
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16, "modnorm_bwd: bf16");
  const long N = x.size(-1);
  TORCH_CHECK(N % 8 == 0 && N > 0 && N <= 16384,
              "modnorm_bwd: bad last dim");
  const long BC = x.numel() / N;
  auto dx = torch::empty_like(x);
  auto dgamma = torch::empty_like(gamma);
  gfa::launch_modnorm_bwd_bf16(ptr<gfa::bf16>(dx), ptr<gfa::bf16>(dgamma),
                               cptr<gfa::bf16>(x), cptr<gfa::bf16>(gamma),
                               cptr<gfa::bf16>(dy), BC, (int)N, (float)eps,
                               cur_stream());
  return {dx, dgamma};
}

Tensor mbstd(Tensor x, int64_t G, int64_t F, double eps) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 4);
  check_dtype(x, "mbstd.x");
  int B = (int)x.size(0), C = (int)x.size(1);
  int HW = (int)(x.size(2) * x.size(3));
  TORCH_CHECK(B % G == 0 && C % F == 0 && G <= 32);
  auto out = torch::empty({B / G, F},
                          x.options().dtype(torch::kFloat32));
  DISPATCH_FT(x, "mbstd", {
    gfa::launch_mbstd<scalar_t>(out.data_ptr<float>(), cptr<scalar_t>(x), B,
                                C, HW, (int)G, (int)F, (float)eps,
                                cur_stream());
  });
  return out;
}

std::vector<Tensor> bipartite_attn_impl(Tensor q, Tensor k, Tensor v,
                                        double scale, bool want_ml) {
  TORCH_CHECK(q.is_cuda() && q.is_contiguous() && q.dim() == 3);
  TORCH_CHECK(k.is_cuda() && k.is_contiguous() && k.dim() == 3);
  TORCH_CHECK(v.is_cuda() && v.is_contiguous() && v.dim() == 3);
  TORCH_CHECK(q.scalar_type() == k.scalar_type() &&
              q.scalar_type() == v.scalar_type());
  check_dtype(q, "bipartite_attn.q");
  gfa::AttnParams p;
  p.B = (int)q.size(0);
  p.Nq = (int)q.size(1);
  p.Nk = (int)k.size(1);
  p.D = (int)q.size(2);
  p.E = (int)v.size(2);
  p.scale = (float)scale;
  TORCH_CHECK(k.size(0) == p.B && v.size(0) == p.B);
  TORCH_CHECK(k.size(2) == p.D && v.size(1) == p.Nk);
  auto out = torch::empty({p.B, p.Nq, p.E}, q.options());
  auto f32opt = q.options().dtype(torch::kFloat32);
  Tensor ml;
  if (p.Nk <= 64) {
    DISPATCH_FT(q, "attn", {
      gfa::launch_attn_smalln<scalar_t>(ptr<scalar_t>(out), cptr<scalar_t>(q),
                                        cptr<scalar_t>(k), cptr<scalar_t>(v),
                                        p, cur_stream());
    });
    // small-N backward recomputes softmax in-kernel: no stats needed
    if (want_ml) ml = torch::empty({0}, f32opt);
  } else {
    TORCH_CHECK(p.Nq <= 64,
                "bipartite_attn: long-N path requires Nq <= 64 "
                "(bipartite attention is k x HW, never HW x HW)");
    int nchunks = (p.Nk + 63) / 64;
    auto ws_m = torch::empty({p.B, nchunks, 64}, f32opt);
    auto ws_l = torch::empty({p.B, nchunks, 64}, f32opt);
    auto ws_o = torch::empty({(int64_t)p.B, nchunks, 64, p.E}, f32opt);
    if (want_ml) ml = torch::empty({p.B, p.Nq, 2}, f32opt);
    DISPATCH_FT(q, "attn", {
      gfa::launch_attn_longn<scalar_t>(
          ptr<scalar_t>(out), cptr<scalar_t>(q), cptr<scalar_t>(k),
          cptr<scalar_t>(v), ws_m.data_ptr<float>(), ws_l.data_ptr<float>(),
          ws_o.data_ptr<float>(),
          want_ml ? ml.data_ptr<float>() : nullptr, p, nchunks,
          cur_stream());
    });
  }
  if (want_ml) return {out, ml};
  return {out};
}

Tensor bipartite_attn(Tensor q, Tensor k, Tensor v, double scale) {
  return bipartite_attn_impl(q, k, v, scale, false)[0];
}

Tensor gemm_skinny(Tensor a, Tensor b, bool trans_b) {
  // C[M,N] = A[M,K] . op(B); trans_b: B is [N,K] (y = x W^T), else [K,N].
  TORCH_CHECK(a.is_cuda() && a.is_contiguous() && a.dim() == 2);
  TORCH_CHECK(b.is_cuda() && b.is_contiguous() && b.dim() == 2);
  TORCH_CHECK(a.scalar_type() == b.scalar_type());
  check_dtype(a, "gemm_skinny.a");
  gfa::GemmParams p;
  p.M = a.size(0);
  p.K = (int)a.size(1);
  if (trans_b) {
    TORCH_CHECK(b.size(1) == p.K, "gemm_skinny: B [N,K] K mismatch");
    p.N = (int)b.size(0);
  } else {
    TORCH_CHECK(b.size(0) == p.K, "gemm_skinny: B [K,N] K mismatch");
    p.N = (int)b.size(1);
  }
  auto c = torch::empty({p.M, p.N}, a.options());
  DISPATCH_FT(a, "gemm_skinny", {
    gfa::launch_gemm_skinny<scalar_t>(ptr<scalar_t>(c), cptr<scalar_t>(a),
                                      cptr<scalar_t>(b), p, trans_b,
                                      cur_stream());
  });
  return c;
}

std::vector<Tensor> bipartite_attn_fwd(Tensor q, Tensor k, Tensor v,
                                       double scale) {
  return bipartite_attn_impl(q, k, v, scale, true);
}

std::vector<Tensor> bipartite_attn_bwd(Tensor q, Tensor k, Tensor v,
                                       Tensor dout, Tensor drow, Tensor ml,
                                       double scale) {
  // drow = (dO * O).sum(-1) fp32 [B,Nq]; ml = [B,Nq,2] fp32 (long-N only).
  TORCH_CHECK(q.is_cuda() && q.is_contiguous() && q.dim() == 3);
  TORCH_CHECK(k.is_cuda() && k.is_contiguous());
  TORCH_CHECK(v.is_cuda() && v.is_contiguous());
  TORCH_CHECK(dout.is_cuda() && dout.is_contiguous() && dout.dim() == 3);
  TORCH_CHECK(drow.is_cuda() && drow.is_contiguous() &&
              drow.scalar_type() == torch::kFloat32);
  TORCH_CHECK(q.scalar_type() == k.scalar_type() &&
              q.scalar_type() == v.scalar_type() &&
              q.scalar_type() == dout.scalar_type());
  check_dtype(q, "bipartite_attn_bwd.q");
  gfa::AttnBwdParams p;
  p.B = (int)q.size(0);
  p.Nq = (int)q.size(1);
  p.Nk = (int)k.size(1);
  p.D = (int)q.size(2);
  p.E = (int)v.size(2);
  p.scale = (float)scale;
  TORCH_CHECK(dout.size(1) == p.Nq && dout.size(2) == p.E);
  TORCH_CHECK(drow.size(0) == p.B && drow.size(1) == p.Nq);
  auto f32opt = q.options().dtype(torch::kFloat32);
  if (p.Nk <= 64) {
    auto dq = torch::empty({p.B, p.Nq, p.D}, q.options());
    auto dkw = torch::zeros({p.B, p.Nk, p.D}, f32opt);
    auto dvw = torch::zeros({p.B, p.Nk, p.E}, f32opt);
    DISPATCH_FT(q, "attn_bwd", {
      gfa::launch_attn_smalln_bwd<scalar_t>(
          ptr<scalar_t>(dq), dkw.data_ptr<float>(), dvw.data_ptr<float>(),
          cptr<scalar_t>(q), cptr<scalar_t>(k), cptr<scalar_t>(v),
          cptr<scalar_t>(dout), drow.data_ptr<float>(), p, cur_stream());
    });
    return {dq, dkw.to(q.scalar_type()), dvw.to(q.scalar_type())};
  }
  TORCH_CHECK(p.Nq <= 64, "bipartite_attn_bwd: long-N needs Nq <= 64");
  TORCH_CHECK(ml.numel() == (int64_t)p.B * p.Nq * 2 && ml.is_cuda() &&
                  ml.is_contiguous() &&
                  ml.scalar_type() == torch::kFloat32,
              "bipartite_attn_bwd: long-N needs the forward's (m,l) stats");
  auto dqw = torch::zeros({p.B, p.Nq, p.D}, f32opt);
  auto dk = torch::empty({p.B, p.Nk, p.D}, q.options());
  auto dv = torch::empty({p.B, p.Nk, p.E}, q.options());
  DISPATCH_FT(q, "attn_bwd", {
    gfa::launch_attn_longn_bwd<scalar_t>(
        dqw.data_ptr<float>(), ptr<scalar_t>(dk), ptr<scalar_t>(dv),
        cptr<scalar_t>(q), cptr<scalar_t>(k), cptr<scalar_t>(v),
        cptr<scalar_t>(dout), ml.data_ptr<float>(), drow.data_ptr<float>(),
        p, cur_stream());
  });
  return {dqw.to(q.scalar_type()), dk, dv};
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "gansformer_amd gfx950 HIP kernels";
  m.def("fba", &fba, "fused bias+act (grad=0) / grad variant (grad=1)");
  m.def("fba_mod", &fba_mod,
        "fused demod-scale + noise + bias + act (synthesis epilogue)");
  m.def("upfirdn2d", &upfirdn2d, "pad-upsample-FIR-downsample");
  m.def("upfirdn2d_sep", &upfirdn2d_sep,
        "separable 4-tap upfirdn (bf16, two-pass LDS)");
  m.def("conv2d_fwd", &conv2d_fwd, "implicit-GEMM conv2d (per-sample ok)");
  m.def("conv2d_wgrad", &conv2d_wgrad, "conv2d weight gradient");
  m.def("conv2d_up2", &conv2d_up2,
        "2x-upsampling 3x3 conv (= conv of zero-stuffed input, pad 1)");
  m.def("mbstd", &mbstd, "minibatch stddev stats [B/G, F]");
  m.def("modnorm", &modnorm,
        "fused instance-norm + modulation over the last dim");
  m.def("modnorm_bwd", &modnorm_bwd,
        "first-order modnorm backward (dx, dgamma)");
  m.def("bipartite_attn", &bipartite_attn, "softmax(QK^T)V");
  m.def("bipartite_attn_fwd", &bipartite_attn_fwd,
        "softmax(QK^T)V returning (out, ml) for the fused backward");
  m.def("bipartite_attn_bwd", &bipartite_attn_bwd,
        "fused attention backward -> (dq, dk, dv)");
  m.def("gemm_skinny", &gemm_skinny,
        "tall-skinny C = A @ op(B) MFMA GEMM");
}
