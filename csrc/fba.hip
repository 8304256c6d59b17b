// Fused bias + activation (+gain +clamp) and its gradient variant.
// Capability parity with the reference's fused_bias_act.cu (ref
// src/dnnlib/tflib/ops/fused_bias_act.cu [R]), redesigned
// for CDNA4: one grid-stride elementwise pass, 8 elements per thread via
// 16-byte vector loads (bf16x8 / f32x4x2) — guide Guideline 13: hipcc
// does not auto-vectorize bf16 scalar loads.
//
// grad == 0 : y = clamp(act(x + b[c]) * gain)
// grad == 1 : dx = dy * slope(y_ref)   (slope reconstructed from the
//             saved forward OUTPUT; zero where |y_ref| hit the clamp)
#include "common.h"

namespace gfa {

// act codes must match gansformer_amd/ops/fused_act.py activation_defs
enum Act { kLinear = 0, kRelu = 1, kLrelu = 2, kTanh = 3, kSigmoid = 4 };

template <int ACT>
GFA_DEV float act_fwd(float v, float alpha) {
  switch (ACT) {
    case kLinear: return v;
    case kRelu: return v > 0.f ? v : 0.f;
    case kLrelu: return v > 0.f ? v : v * alpha;
    case kTanh: return tanhf(v);
    case kSigmoid: return 1.f / (1.f + __expf(-v));
  }
  return v;
}

// slope dy/dx expressed through the saved output y (y includes gain).
template <int ACT>
GFA_DEV float act_slope_from_y(float y, float alpha, float gain) {
  switch (ACT) {
    case kLinear: return gain;
    case kRelu: return y > 0.f ? gain : 0.f;
    case kLrelu: return y > 0.f ? gain : gain * alpha;
    case kTanh: { float t = y / gain; return gain * (1.f - t * t); }
    case kSigmoid: { float s = y / gain; return gain * s * (1.f - s); }
  }
  return gain;
}

template <typename T, int ACT, bool HAS_BIAS, int GRAD>
__global__ void fba_kernel(T* __restrict__ out_, const T* __restrict__ x_,
                           const float* __restrict__ b,
                           const T* __restrict__ ref_, long n, long inner,
                           int C, float alpha, float gain, float clamp) {
  using R = typename Raw<T>::type;   // bit-level element (u16 for bf16)
  constexpr int V = 16 / sizeof(T);  // elements per 16B vector
  typedef R vec_t __attribute__((ext_vector_type(V)));
  R* out = reinterpret_cast<R*>(out_);
  const R* x = reinterpret_cast<const R*>(x_);
  const R* ref = reinterpret_cast<const R*>(ref_);
  long i0 = global_tid() * V;
  long stride = global_stride() * V;
  for (long i = i0; i < n; i += stride) {
    if (i + V <= n) {
      vec_t xv = *reinterpret_cast<const vec_t*>(x + i);
      vec_t rv = {};
      if (GRAD == 1) rv = *reinterpret_cast<const vec_t*>(ref + i);
      vec_t yv;
#pragma unroll
      for (int j = 0; j < V; ++j) {
        float v = raw_to_f32((R)xv[j]);
        if (GRAD == 0) {
          if (HAS_BIAS) v += b[((i + j) / inner) % C];
          float y = act_fwd<ACT>(v, alpha) * gain;
          y = fminf(fmaxf(y, -clamp), clamp);
          yv[j] = f32_to_raw<R>(y);
        } else {
          float yr = raw_to_f32((R)rv[j]);
          float dx = v * act_slope_from_y<ACT>(yr, alpha, gain);
          if (fabsf(yr) >= clamp) dx = 0.f;
          yv[j] = f32_to_raw<R>(dx);
        }
      }
      *reinterpret_cast<vec_t*>(out + i) = yv;
    } else {
      for (long j = i; j < n; ++j) {
        float v = raw_to_f32(x[j]);
        if (GRAD == 0) {
          if (HAS_BIAS) v += b[(j / inner) % C];
          float y = act_fwd<ACT>(v, alpha) * gain;
          out[j] = f32_to_raw<R>(fminf(fmaxf(y, -clamp), clamp));
        } else {
          float yr = raw_to_f32(ref[j]);
          float dx = v * act_slope_from_y<ACT>(yr, alpha, gain);
          if (fabsf(yr) >= clamp) dx = 0.f;
          out[j] = f32_to_raw<R>(dx);
        }
      }
    }
  }
}

template <typename T>
void launch_fba(T* out, const T* x, const float* b, const T* ref, long n,
                long inner, int C, int act, int grad, float alpha, float gain,
                float clamp, hipStream_t stream) {
  constexpr int V = 16 / sizeof(T);
  dim3 grid = stream_grid((n + V - 1) / V);
  bool has_bias = (b != nullptr) && grad == 0;
#define GFA_DISPATCH(ACTC)                                                   \
  do {                                                                       \
    if (grad == 0) {                                                         \
      if (has_bias)                                                          \
        hipLaunchKernelGGL((fba_kernel<T, ACTC, true, 0>), grid, dim3(256),  \
                           0, stream, out, x, b, ref, n, inner, C, alpha,    \
                           gain, clamp);                                     \
      else                                                                   \
        hipLaunchKernelGGL((fba_kernel<T, ACTC, false, 0>), grid, dim3(256), \
                           0, stream, out, x, b, ref, n, inner, C, alpha,    \
                           gain, clamp);                                     \
    } else {                                                                 \
      hipLaunchKernelGGL((fba_kernel<T, ACTC, false, 1>), grid, dim3(256),   \
                         0, stream, out, x, b, ref, n, inner, C, alpha,      \
                         gain, clamp);                                       \
    }                                                                        \
  } while (0)
  switch (act) {
    case kLinear: GFA_DISPATCH(kLinear); break;
    case kRelu: GFA_DISPATCH(kRelu); break;
    case kLrelu: GFA_DISPATCH(kLrelu); break;
    case kTanh: GFA_DISPATCH(kTanh); break;
    case kSigmoid: GFA_DISPATCH(kSigmoid); break;
    default: break;
  }
#undef GFA_DISPATCH
}

template void launch_fba<float>(float*, const float*, const float*,
                                const float*, long, long, int, int, int,
                                float, float, float, hipStream_t);
template void launch_fba<bf16>(bf16*, const bf16*, const float*, const bf16*,
                               long, long, int, int, int, float, float, float,
                               hipStream_t);


// ---- fused demod-scale + noise + bias + act (SynthesisLayer epilogue) ----
// y = clamp(act(x * d[b,c] + noise[b,hw] * sigma + bias[c]) * gain)
// Folds the three elementwise passes that followed every modulated conv
// (demodulation scale, noise add, bias+act) into one.  bf16, NCHW,
// HW % 8 == 0.
template <int ACT, bool HAS_NOISE>
__global__ void fba_mod_kernel(bf16* __restrict__ out_,
                               const bf16* __restrict__ x_,
                               const float* __restrict__ d,   // [B*C]
                               const bf16* __restrict__ noise_,  // [B*HW]
                               const float* __restrict__ bias,   // [C]
                               const float* __restrict__ sigma_p,  // [1]
                               long n, long hw, int C,
                               float alpha, float gain, float clamp) {
  const float sigma = HAS_NOISE ? sigma_p[0] : 0.f;
  typedef u16 vec_t __attribute__((ext_vector_type(8)));
  u16* out = reinterpret_cast<u16*>(out_);
  const u16* x = reinterpret_cast<const u16*>(x_);
  const u16* noise = reinterpret_cast<const u16*>(noise_);
  const long i0 = global_tid() * 8;
  const long stride = global_stride() * 8;
  for (long i = i0; i < n; i += stride) {
    const long bc = i / hw;
    const long b = bc / C;
    const float dv = d[bc];
    const float bv = bias ? bias[bc % C] : 0.f;
    const long pos = i - bc * hw;
    vec_t xv = *reinterpret_cast<const vec_t*>(x + i);
    vec_t nv = {};
    if (HAS_NOISE)
      nv = *reinterpret_cast<const vec_t*>(noise + b * hw + pos);
    vec_t yv;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float v = bf16_bits_to_f32(xv[j]) * dv + bv;
      if (HAS_NOISE) v += bf16_bits_to_f32(nv[j]) * sigma;
      float y = act_fwd<ACT>(v, alpha) * gain;
      yv[j] = f32_to_bf16_bits(fminf(fmaxf(y, -clamp), clamp));
    }
    *reinterpret_cast<vec_t*>(out + i) = yv;
  }
}

void launch_fba_mod(bf16* out, const bf16* x, const float* d,
                    const bf16* noise, const float* bias,
                    const float* sigma, long n, long hw, int C, int act,
                    float alpha, float gain, float clamp,
                    hipStream_t stream) {
  dim3 grid = stream_grid(n / 8);
#define GFA_MOD_DISPATCH(ACTC)                                              \
  do {                                                                      \
    if (noise)                                                              \
      hipLaunchKernelGGL((fba_mod_kernel<ACTC, true>), grid, dim3(256), 0,  \
                         stream, out, x, d, noise, bias, sigma, n, hw, C,   \
                         alpha, gain, clamp);                               \
    else                                                                    \
      hipLaunchKernelGGL((fba_mod_kernel<ACTC, false>), grid, dim3(256), 0, \
                         stream, out, x, d, noise, bias, sigma, n, hw, C,   \
                         alpha, gain, clamp);                               \
  } while (0)
  switch (act) {
    case kLinear: GFA_MOD_DISPATCH(kLinear); break;
    case kLrelu: GFA_MOD_DISPATCH(kLrelu); break;
    default: break;
  }
#undef GFA_MOD_DISPATCH
}

}  // namespace gfa
