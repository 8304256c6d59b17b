// Fused spatial instance-norm + attention modulation for gfx950.
//
//   y[b,c,t] = (x[b,c,t] - mu[b,c]) * rstd[b,c] * (1 + gamma[b,c,t])
//              + beta[b,c,t]
//
// This is the GANsformer's "mul" integration (attention output modulates
// the feature map's mean/var, ref src/training/networks.py [R], SURVEY.md
// M3): the eager path spent ~10 full-tensor fp32 passes (float cast,
// mean, var, normalize, scale, shift, cast back, permute). Here one
// workgroup owns one (b, c) row of N = H*W bf16 elements: a vectorized
// stats pass (f32 accumulation, LDS tree reduce) and a modulate pass.
// mean/rstd are returned for the autograd backward (composed of plain
// torch ops, so R1/path-length double-backward stays exact).
#include "common.h"

namespace gfa {

__global__ __launch_bounds__(256)
void modnorm_fwd_bf16(bf16* __restrict__ y, float* __restrict__ mean,
                      float* __restrict__ rstd, const bf16* __restrict__ x,
                      const bf16* __restrict__ gamma,
                      const bf16* __restrict__ beta, long BC, int N,
                      float eps) {
  __shared__ float red[2][256];

  for (long bc = blockIdx.x; bc < BC; bc += gridDim.x) {
    const u16* xr = reinterpret_cast<const u16*>(x) + bc * N;
    const int t = threadIdx.x;

    float s = 0.f, s2 = 0.f;
    for (int i = t * 8; i < N; i += 256 * 8) {
      s16x8 v8 = *reinterpret_cast<const s16x8*>(xr + i);
      const u16* v = reinterpret_cast<const u16*>(&v8);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float v32 = bf16_bits_to_f32(v[j]);
        s += v32;
        s2 += v32 * v32;
      }
    }
    red[0][t] = s;
    red[1][t] = s2;
    __syncthreads();
    for (int w = 128; w > 0; w >>= 1) {
      if (t < w) {
        red[0][t] += red[0][t + w];
        red[1][t] += red[1][t + w];
      }
      __syncthreads();
    }
    const float mu = red[0][0] / N;
    const float var = fmaxf(red[1][0] / N - mu * mu, 0.f);
    const float rs = __frsqrt_rn(var + eps);
    if (t == 0) {
      mean[bc] = mu;
      rstd[bc] = rs;
    }

    const u16* gr = reinterpret_cast<const u16*>(gamma) + bc * N;
    const u16* br = reinterpret_cast<const u16*>(beta) + bc * N;
    u16* yr = reinterpret_cast<u16*>(y) + bc * N;
    for (int i = t * 8; i < N; i += 256 * 8) {
      s16x8 v8 = *reinterpret_cast<const s16x8*>(xr + i);
      s16x8 g8 = *reinterpret_cast<const s16x8*>(gr + i);
      s16x8 b8 = *reinterpret_cast<const s16x8*>(br + i);
      const u16* v = reinterpret_cast<const u16*>(&v8);
      const u16* g = reinterpret_cast<const u16*>(&g8);
      const u16* b = reinterpret_cast<const u16*>(&b8);
      u16 o[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float xn = (bf16_bits_to_f32(v[j]) - mu) * rs;
        o[j] = f32_to_bf16_bits(
            xn * (1.f + bf16_bits_to_f32(g[j])) + bf16_bits_to_f32(b[j]));
      }
      *reinterpret_cast<s16x8*>(yr + i) = *reinterpret_cast<const s16x8*>(o);
    }
    __syncthreads();
  }
}

void launch_modnorm_fwd_bf16(bf16* y, float* mean, float* rstd,
                             const bf16* x, const bf16* gamma,
                             const bf16* beta, long BC, int N, float eps,
                             hipStream_t s) {
  long grid = BC < 4096 ? BC : 4096;
  hipLaunchKernelGGL(modnorm_fwd_bf16, dim3((unsigned)grid), dim3(256), 0, s,
                     y, mean, rstd, x, gamma, beta, BC, N, eps);
}

// First-order backward (x, gamma, dy -> dx, dgamma; dbeta == dy at the
// Python layer). Instance-norm backward per (b, c) row:
//   xn  = (x - mu) * rstd
//   dxn = dy * (1 + gamma)            dgamma = dy * xn
//   dx  = rstd * (dxn - mean(dxn) - xn * mean(dxn * xn))
// x and dxn are stashed in LDS between the passes, so global traffic is
// 3 reads + 2 writes total (the eager composition was ~12 passes).
// Path-length double-backward does NOT come through here: the Python
// backward uses the differentiable eager composition when grad mode is
// on (create_graph replays).
__global__ __launch_bounds__(256)
void modnorm_bwd_bf16(bf16* __restrict__ dx, bf16* __restrict__ dgamma,
                      const bf16* __restrict__ x,
                      const bf16* __restrict__ gamma,
                      const bf16* __restrict__ dy, long BC, int N,
                      float eps) {
  extern __shared__ u16 sm[];  // [N] x row, then [N] dxn row
  __shared__ float red[2][256];
  u16* sx = sm;
  u16* sdxn = sm + N;

  for (long bc = blockIdx.x; bc < BC; bc += gridDim.x) {
    const u16* xr = reinterpret_cast<const u16*>(x) + bc * N;
    const u16* gr = reinterpret_cast<const u16*>(gamma) + bc * N;
    const u16* dr = reinterpret_cast<const u16*>(dy) + bc * N;
    const int t = threadIdx.x;

    // pass 1: stats over x, stash x in LDS
    float s = 0.f, s2 = 0.f;
    for (int i = t * 8; i < N; i += 256 * 8) {
      s16x8 v8 = *reinterpret_cast<const s16x8*>(xr + i);
      *reinterpret_cast<s16x8*>(sx + i) = v8;
      const u16* v = reinterpret_cast<const u16*>(&v8);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float v32 = bf16_bits_to_f32(v[j]);
        s += v32;
        s2 += v32 * v32;
      }
    }
    red[0][t] = s;
    red[1][t] = s2;
    __syncthreads();
    for (int w = 128; w > 0; w >>= 1) {
      if (t < w) {
        red[0][t] += red[0][t + w];
        red[1][t] += red[1][t + w];
      }
      __syncthreads();
    }
    const float mu = red[0][0] / N;
    const float var = fmaxf(red[1][0] / N - mu * mu, 0.f);
    const float rs = __frsqrt_rn(var + eps);
    __syncthreads();

    // pass 2: dxn + dgamma, accumulate sum(dxn), sum(dxn*xn)
    float s1 = 0.f, sxn = 0.f;
    for (int i = t * 8; i < N; i += 256 * 8) {
      s16x8 g8 = *reinterpret_cast<const s16x8*>(gr + i);
      s16x8 d8 = *reinterpret_cast<const s16x8*>(dr + i);
      const u16* g = reinterpret_cast<const u16*>(&g8);
      const u16* d = reinterpret_cast<const u16*>(&d8);
      u16 og[8], on[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float xn = (bf16_bits_to_f32(sx[i + j]) - mu) * rs;
        const float dyv = bf16_bits_to_f32(d[j]);
        const float dxn = dyv * (1.f + bf16_bits_to_f32(g[j]));
        og[j] = f32_to_bf16_bits(dyv * xn);
        on[j] = f32_to_bf16_bits(dxn);
        s1 += dxn;
        sxn += dxn * xn;
      }
      *reinterpret_cast<s16x8*>(
          reinterpret_cast<u16*>(dgamma) + bc * N + i) =
          *reinterpret_cast<const s16x8*>(og);
      *reinterpret_cast<s16x8*>(sdxn + i) =
          *reinterpret_cast<const s16x8*>(on);
    }
    red[0][t] = s1;
    red[1][t] = sxn;
    __syncthreads();
    for (int w = 128; w > 0; w >>= 1) {
      if (t < w) {
        red[0][t] += red[0][t + w];
        red[1][t] += red[1][t + w];
      }
      __syncthreads();
    }
    const float m1 = red[0][0] / N;
    const float m2 = red[1][0] / N;

    // pass 3: dx from LDS only
    for (int i = t * 8; i < N; i += 256 * 8) {
      u16 o[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float xn = (bf16_bits_to_f32(sx[i + j]) - mu) * rs;
        const float dxn = bf16_bits_to_f32(sdxn[i + j]);
        o[j] = f32_to_bf16_bits(rs * (dxn - m1 - xn * m2));
      }
      *reinterpret_cast<s16x8*>(reinterpret_cast<u16*>(dx) + bc * N + i) =
          *reinterpret_cast<const s16x8*>(o);
    }
    __syncthreads();
  }
}

void launch_modnorm_bwd_bf16(bf16* dx, bf16* dgamma, const bf16* x,
                             const bf16* gamma, const bf16* dy, long BC,
                             int N, float eps, hipStream_t s) {
  long grid = BC < 4096 ? BC : 4096;
  const size_t lds = (size_t)2 * N * sizeof(u16);
  hipLaunchKernelGGL(modnorm_bwd_bf16, dim3((unsigned)grid), dim3(256), lds,
                     s, dx, dgamma, x, gamma, dy, BC, N, eps);
}

}  // namespace gfa
