#include "hip/hip_runtime.h"
// Implicit-GEMM conv2d on MFMA for gfx950, with native per-sample weights.
// Replaces the reference's cuDNN convs + grouped-conv modulation trick
// (ref src/training/networks.py modulated conv).
//
// Forward GEMM view (per sample b):
//   A = W[b?]           [O][K]   K = I*kh*kw   (contiguous rows)
//   B = X patches       [K][P]   implicit im2col addressing
//   C = Y[b]            [O][P]
// Tile: 64(O) x 64(P) x BK, 4 waves (2x2), each wave a 32x32 sub-tile of
// 2x2 MFMA fragments. A and B~ tiles staged in LDS with an 8-element XOR
// swizzle so the 16-byte fragment reads are bank-spread (guide T2).
//
// Weight-gradient GEMM view (wgrad):
//   A = dY[b]           [O][P]
//   B = X patches       [Kw][P]  Kw = I*kh*kw
//   C = dW[b?]          [O][Kw]  (shared weights: sum over b in the K loop)
//
// bf16 path: v_mfma_f32_16x16x32_bf16.  f32 path: v_mfma_f32_16x16x4_f32
// (exact f32 at the f32 vector rate — there is no xf32 on gfx950).
#include "common.h"

namespace gfa {

struct ConvParams {
  int B, I, H, W;       // input
  int O, OH, OW;        // output
  int kh, kw, stride, pad;
  int per_sample;       // weights have a leading B dim
};

// ---------------- bf16 forward ----------------
// LDS tiles: As[64][32] bf16 (A rows), Bs[64][32] bf16 (pixel rows, k cols)
// swizzle: 8-element granule g at row r stored at g ^ (r & 3).

__device__ __forceinline__ int swz(int row, int g) { return g ^ (row & 3); }

__global__ __launch_bounds__(256)
void conv2d_fwd_bf16(bf16* __restrict__ y, const bf16* __restrict__ x,
                     const bf16* __restrict__ w, ConvParams p) {
  constexpr int BM = 64, BN = 64, BK = 32, RP = 40;  // padded row (bank-safe)
  __shared__ u16 As[BM * RP];
  __shared__ u16 Bs[BN * RP];

  const int K = p.I * p.kh * p.kw;
  const int P = p.OH * p.OW;
  const int b = blockIdx.z;
  const int m0 = blockIdx.y * BM;
  const int n0 = blockIdx.x * BN;
  const bf16* wb = w + (p.per_sample ? (long)b * p.O * K : 0);
  const bf16* xb = x + (long)b * p.I * p.H * p.W;
  bf16* yb = y + (long)b * p.O * P;

  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const int wm = (wave >> 1) * 32;  // wave row offset in tile
  const int wn = (wave & 1) * 32;

  f32x4 acc[2][2] = {};
  const bool a_vec = (K % 8 == 0);

  for (int k0 = 0; k0 < K; k0 += BK) {
    // ---- stage A: 64 rows x 32 k, thread t -> row t/4, granule t%4
    {
      int row = t >> 2, g = t & 3;
      int o = m0 + row;
      int kk = k0 + g * 8;
      u16* dst = &As[row * RP + g * 8];
      if (o < p.O && kk + 8 <= K && a_vec) {
        *reinterpret_cast<s16x8*>(dst) =
            *reinterpret_cast<const s16x8*>(wb + (long)o * K + kk);
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          u16 v = 0;
          if (o < p.O && kk + j < K)
            v = __builtin_bit_cast(u16, wb[(long)o * K + kk + j]);
          dst[j] = v;
        }
      }
    }
    // ---- stage B: 64 pixels x 32 k; pass over k-slices, coalesced in p
#pragma unroll
    for (int pass = 0; pass < 8; ++pass) {
      int kk = k0 + pass * 4 + (t >> 6);  // 4 k per pass x 4 waves
      int pix = n0 + (t & 63);
      u16 v = 0;
      if (kk < K && pix < P) {
        int i = kk / (p.kh * p.kw);
        int rs = kk % (p.kh * p.kw);
        int r = rs / p.kw, s = rs % p.kw;
        int oy = pix / p.OW, ox = pix % p.OW;
        int iy = oy * p.stride + r - p.pad;
        int ix = ox * p.stride + s - p.pad;
        if (iy >= 0 && iy < p.H && ix >= 0 && ix < p.W)
          v = __builtin_bit_cast(u16, xb[((long)i * p.H + iy) * p.W + ix]);
      }
      int lk = kk - k0;
      int lp = pix - n0;
      Bs[lp * RP + lk] = v;
    }
    __syncthreads();
    // ---- MFMA: each wave 2x2 fragments of 16x16, K=32 in one mfma each
#pragma unroll
    for (int mi = 0; mi < 2; ++mi) {
#pragma unroll
      for (int ni = 0; ni < 2; ++ni) {
        int arow = wm + mi * 16 + (lane & 15);
        int ag = lane >> 4;
        s16x8 af = *reinterpret_cast<const s16x8*>(
            &As[arow * RP + ag * 8]);
        int brow = wn + ni * 16 + (lane & 15);
        s16x8 bf = *reinterpret_cast<const s16x8*>(
            &Bs[brow * RP + ag * 8]);
        acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            af, bf, acc[mi][ni], 0, 0, 0);
      }
    }
    __syncthreads();
  }
  // ---- epilogue: C/D map col=lane&15, row=(lane>>4)*4+reg
#pragma unroll
  for (int mi = 0; mi < 2; ++mi)
#pragma unroll
    for (int ni = 0; ni < 2; ++ni) {
      int col = n0 + wn + ni * 16 + (lane & 15);
      if (col >= P) continue;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        int row = m0 + wm + mi * 16 + (lane >> 4) * 4 + reg;
        if (row < p.O)
          yb[(long)row * P + col] = from_f32<bf16>(acc[mi][ni][reg]);
      }
    }
}

// ---------------- bf16 forward, 128x128 tile ----------------
// For large pixel counts: 4 waves, each owning a 64x64 sub-tile (4x4
// fragments), BK=32. 16 MFMA per 8 fragment ds_read_b128 per wave per
// K-step — 4x the MFMA:staging ratio of the 64x64 kernel.
__global__ __launch_bounds__(256)
void conv2d_fwd_bf16_128(bf16* __restrict__ y, const bf16* __restrict__ x,
                         const bf16* __restrict__ w, ConvParams p) {
  constexpr int BM = 128, BN = 128, BK = 32, RP = 40;
  __shared__ u16 As[BM * RP];
  __shared__ u16 Bs[BN * RP];

  const int K = p.I * p.kh * p.kw;
  const int P = p.OH * p.OW;
  const int b = blockIdx.z;
  const int m0 = blockIdx.y * BM;
  const int n0 = blockIdx.x * BN;
  const bf16* wb = w + (p.per_sample ? (long)b * p.O * K : 0);
  const bf16* xb = x + (long)b * p.I * p.H * p.W;
  bf16* yb = y + (long)b * p.O * P;

  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const int wm = (wave >> 1) * 64;
  const int wn = (wave & 1) * 64;
  const int khw = p.kh * p.kw;

  // precompute this thread's two staged pixels' base coords (fixed for
  // the whole K loop — kills the per-element div/mod that VALU-bound v1)
  int pix_a = n0 + lane, pix_b = n0 + 64 + lane;
  int iya = (pix_a / p.OW) * p.stride - p.pad;
  int ixa = (pix_a % p.OW) * p.stride - p.pad;
  int iyb = (pix_b / p.OW) * p.stride - p.pad;
  int ixb = (pix_b % p.OW) * p.stride - p.pad;
  const bool va = pix_a < P, vb = pix_b < P;

  // incremental tap state (i, r, s) for this wave's first k of the step —
  // advanced without division inside the K loop (divisions are VALU
  // poison: kk/khw per lane per element made v2 VALU-bound).
  int ti = (wave * 8) / khw;
  int trs = (wave * 8) % khw;
  int tr = trs / p.kw, ts = trs % p.kw;
  const int q24 = 24 / khw, r24 = 24 % khw;  // advance by 24 = 32 - 8

  f32x4 acc[4][4] = {};
  const bool a_vec = (K % 8 == 0);

  for (int k0 = 0; k0 < K; k0 += BK) {
    // ---- stage A: 128 rows x 32 k; thread t covers rows t/4 and 64+t/4
#pragma unroll
    for (int half = 0; half < 2; ++half) {
      int row = half * 64 + (t >> 2), g = t & 3;
      int o = m0 + row;
      int kk = k0 + g * 8;
      u16* dst = &As[row * RP + (g << 3)];
      if (o < p.O && kk + 8 <= K && a_vec) {
        *reinterpret_cast<s16x8*>(dst) =
            *reinterpret_cast<const s16x8*>(wb + (long)o * K + kk);
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          u16 v = 0;
          if (o < p.O && kk + j < K)
            v = __builtin_bit_cast(u16, wb[(long)o * K + kk + j]);
          dst[j] = v;
        }
      }
    }
    // ---- stage B: 128 pixels x 32 k; wave w owns k range [w*8, w*8+8),
    // lanes sweep pixels (coalesced 64-wide); incremental tap decode.
    {
      int i = ti, r = tr, s = ts;
#pragma unroll
      for (int kj = 0; kj < 8; ++kj) {
        int lk = wave * 8 + kj;
        int kk = k0 + lk;
        u16 va16 = 0, vb16 = 0;
        if (kk < K) {
          const bf16* xi = xb + (long)i * p.H * p.W;
          int ya = iya + r, xa = ixa + s;
          if (va && ya >= 0 && ya < p.H && xa >= 0 && xa < p.W)
            va16 = __builtin_bit_cast(u16, xi[(long)ya * p.W + xa]);
          int yb2 = iyb + r, xb2 = ixb + s;
          if (vb && yb2 >= 0 && yb2 < p.H && xb2 >= 0 && xb2 < p.W)
            vb16 = __builtin_bit_cast(u16, xi[(long)yb2 * p.W + xb2]);
        }
        Bs[lane * RP + lk] = va16;
        Bs[(64 + lane) * RP + lk] = vb16;
        // advance (i, r, s) by one tap
        if (++s == p.kw) {
          s = 0;
          if (++r == p.kh) { r = 0; ++i; }
        }
      }
      // advance the wave state by the remaining 24 to reach k0 + BK
      ti = i + q24;
      tr = r;
      ts = s;
      int rs = tr * p.kw + ts + r24;
      if (rs >= khw) { rs -= khw; ++ti; }
      tr = rs / 3;  // kw<=3 in this framework; exact for kw==3
      ts = rs - tr * 3;
      if (p.kw != 3) { tr = rs / p.kw; ts = rs % p.kw; }
    }
    __syncthreads();
#pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
      int arow = wm + mi * 16 + (lane & 15);
      int ag = lane >> 4;
      s16x8 af = *reinterpret_cast<const s16x8*>(
          &As[arow * RP + (ag << 3)]);
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        int brow = wn + ni * 16 + (lane & 15);
        s16x8 bfr = *reinterpret_cast<const s16x8*>(
            &Bs[brow * RP + (ag << 3)]);
        acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            af, bfr, acc[mi][ni], 0, 0, 0);
      }
    }
    __syncthreads();
  }
#pragma unroll
  for (int mi = 0; mi < 4; ++mi)
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      int col = n0 + wn + ni * 16 + (lane & 15);
      if (col >= P) continue;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        int row = m0 + wm + mi * 16 + (lane >> 4) * 4 + reg;
        if (row < p.O)
          yb[(long)row * P + col] = from_f32<bf16>(acc[mi][ni][reg]);
      }
    }
}

// ---------------- f32 forward (exact f32, 16x16x4 MFMA) ----------------
__global__ __launch_bounds__(256)
void conv2d_fwd_f32(float* __restrict__ y, const float* __restrict__ x,
                    const float* __restrict__ w, ConvParams p) {
  constexpr int BM = 64, BN = 64, BK = 16;
  __shared__ float As[BM][BK + 1];
  __shared__ float Bs[BN][BK + 1];

  const int K = p.I * p.kh * p.kw;
  const int P = p.OH * p.OW;
  const int b = blockIdx.z;
  const int m0 = blockIdx.y * BM;
  const int n0 = blockIdx.x * BN;
  const float* wb = w + (p.per_sample ? (long)b * p.O * K : 0);
  const float* xb = x + (long)b * p.I * p.H * p.W;
  float* yb = y + (long)b * p.O * P;

  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const int wm = (wave >> 1) * 32;
  const int wn = (wave & 1) * 32;

  f32x4 acc[2][2] = {};

  for (int k0 = 0; k0 < K; k0 += BK) {
    {  // stage A: thread t -> row t/4, k-chunk of 4
      int row = t >> 2, c4 = (t & 3) * 4;
      int o = m0 + row;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        int kk = k0 + c4 + j;
        As[row][c4 + j] =
            (o < p.O && kk < K) ? wb[(long)o * K + kk] : 0.f;
      }
    }
#pragma unroll
    for (int pass = 0; pass < 4; ++pass) {
      int kk = k0 + pass * 4 + (t >> 6);
      int pix = n0 + (t & 63);
      float v = 0.f;
      if (kk < K && pix < P) {
        int i = kk / (p.kh * p.kw);
        int rs = kk % (p.kh * p.kw);
        int r = rs / p.kw, s = rs % p.kw;
        int oy = pix / p.OW, ox = pix % p.OW;
        int iy = oy * p.stride + r - p.pad;
        int ix = ox * p.stride + s - p.pad;
        if (iy >= 0 && iy < p.H && ix >= 0 && ix < p.W)
          v = xb[((long)i * p.H + iy) * p.W + ix];
      }
      Bs[pix - n0][kk - k0] = v;
    }
    __syncthreads();
#pragma unroll
    for (int kk = 0; kk < BK; kk += 4) {
#pragma unroll
      for (int mi = 0; mi < 2; ++mi)
#pragma unroll
        for (int ni = 0; ni < 2; ++ni) {
          float av = As[wm + mi * 16 + (lane & 15)][kk + (lane >> 4)];
          float bv = Bs[wn + ni * 16 + (lane & 15)][kk + (lane >> 4)];
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x4f32(
              av, bv, acc[mi][ni], 0, 0, 0);
        }
    }
    __syncthreads();
  }
#pragma unroll
  for (int mi = 0; mi < 2; ++mi)
#pragma unroll
    for (int ni = 0; ni < 2; ++ni) {
      int col = n0 + wn + ni * 16 + (lane & 15);
      if (col >= P) continue;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        int row = m0 + wm + mi * 16 + (lane >> 4) * 4 + reg;
        if (row < p.O) yb[(long)row * P + col] = acc[mi][ni][reg];
      }
    }
}

// ---------------- wgrad ----------------
// dW[b?][o][kw_idx] = sum_p dY[b][o][p] * patch[b][kw_idx][p]
// GEMM: M = O (dY rows, p-contiguous -> 16B vector staging), N = Kw
// (implicit patches, 16B staging when the 8-pixel run is contiguous and
// in-bounds), K = P (x B for shared weights). MFMA via TileOps: bf16 path
// on v_mfma_f32_16x16x32_bf16, f32 on the exact-f32 16x16x4.
template <typename T>
__global__ __launch_bounds__(256)
void conv2d_wgrad_kernel(T* __restrict__ dw, float* __restrict__ ws,
                         const T* __restrict__ x, const T* __restrict__ dy,
                         ConvParams p, int nsplit) {
  using TO = TileOps<T>;
  constexpr int BK = TO::BK;   // pixels per K step (32)
  constexpr int ROW = TO::ROW;
  using elem = typename TO::elem;
  __shared__ elem As[64 * ROW];
  __shared__ elem Bs[64 * ROW];

  const int Kw = p.I * p.kh * p.kw;
  const int P = p.OH * p.OW;
  const int m0 = blockIdx.y * 64;      // O tile
  const int n0 = blockIdx.x * 64;      // Kw tile
  const int nb = p.per_sample ? 1 : p.B;
  const int b_fix = p.per_sample ? (int)blockIdx.z / nsplit : 0;
  // split-K: this block covers K-iterations [it0, it1) of nb * ceil(P/BK)
  const int split = p.per_sample ? (int)blockIdx.z % nsplit : (int)blockIdx.z;
  const int iters_per_b = (P + BK - 1) / BK;
  const long total_iters = (long)nb * iters_per_b;
  const long span = (total_iters + nsplit - 1) / nsplit;
  const long it0 = split * span;
  const long it1 = min(total_iters, it0 + span);

  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const int wm = (wave >> 1) * 32;
  const int wn = (wave & 1) * 32;

  // per-thread staging roles (constant over the K loop)
  const int srow = t >> 2;             // LDS row this thread fills
  const int sg = t & 3;                // granule (8 K-elements)
  const int b_kwi = n0 + srow;
  const int b_i = b_kwi / (p.kh * p.kw);
  const int b_rs = b_kwi % (p.kh * p.kw);
  const int b_r = b_rs / p.kw, b_s = b_rs % p.kw;

  f32x4 acc[2][2] = {};

  for (long it = it0; it < it1; ++it) {
    const int bb = (int)(it / iters_per_b);
    const int p0 = (int)(it % iters_per_b) * BK;
    const int b = p.per_sample ? b_fix : bb;
    const T* dyb = dy + (long)b * p.O * P;
    const T* xb = x + (long)b * p.I * p.H * p.W;
    {
      {  // ---- stage A (dY): always p-contiguous in memory
        int o = m0 + srow;
        int pp = p0 + sg * 8;
        if constexpr (sizeof(T) == 2) {
          if (o < p.O && pp + 8 <= P && (((long)o * P + pp) & 7) == 0) {
            TO::store_vec8(As, srow, sg,
                           *reinterpret_cast<const s16x8*>(
                               reinterpret_cast<const u16*>(dyb) +
                               (long)o * P + pp));
          } else {
#pragma unroll
            for (int j = 0; j < 8; ++j)
              TO::store(As, srow, sg * 8 + j,
                        (o < p.O && pp + j < P)
                            ? to_f32(dyb[(long)o * P + pp + j]) : 0.f);
          }
        } else {
#pragma unroll
          for (int j = 0; j < 8; ++j)
            TO::store(As, srow, sg * 8 + j,
                      (o < p.O && pp + j < P)
                          ? to_f32(dyb[(long)o * P + pp + j]) : 0.f);
        }
      }
      {  // ---- stage B (patches): vector when the run is one x row
        int pp = p0 + sg * 8;
        int oy0 = pp / p.OW, ox0 = pp % p.OW;
        bool fast = false;
        long src = 0;
        if (b_kwi < Kw && pp + 7 < P && p.stride == 1 &&
            (pp + 7) / p.OW == oy0) {
          int iy = oy0 + b_r - p.pad;
          int ix = ox0 + b_s - p.pad;
          if (iy >= 0 && iy < p.H && ix >= 0 && ix + 7 < p.W) {
            fast = true;
            src = ((long)b_i * p.H + iy) * p.W + ix;
          }
        }
        if constexpr (sizeof(T) == 2) {
          if (fast && (src & 7) == 0) {
            TO::store_vec8(Bs, srow, sg,
                           *reinterpret_cast<const s16x8*>(
                               reinterpret_cast<const u16*>(xb) + src));
            fast = true;
          } else {
            fast = false;
          }
        } else {
          fast = false;
        }
        if (!fast) {
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            int pj = pp + j;
            float v = 0.f;
            if (b_kwi < Kw && pj < P) {
              int oy = pj / p.OW, ox = pj % p.OW;
              int iy = oy * p.stride + b_r - p.pad;
              int ix = ox * p.stride + b_s - p.pad;
              if (iy >= 0 && iy < p.H && ix >= 0 && ix < p.W)
                v = to_f32(xb[((long)b_i * p.H + iy) * p.W + ix]);
            }
            TO::store(Bs, srow, sg * 8 + j, v);
          }
        }
      }
      __syncthreads();
#pragma unroll
      for (int mi = 0; mi < 2; ++mi)
#pragma unroll
        for (int ni = 0; ni < 2; ++ni)
          acc[mi][ni] = TO::mfma(As, Bs, wm + mi * 16, wn + ni * 16, lane,
                                 acc[mi][ni]);
      __syncthreads();
    }
  }
  const long wbase = p.per_sample ? (long)b_fix * p.O * Kw : 0;
#pragma unroll
  for (int mi = 0; mi < 2; ++mi)
#pragma unroll
    for (int ni = 0; ni < 2; ++ni) {
      int col = n0 + wn + ni * 16 + (lane & 15);
      if (col >= Kw) continue;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        int row = m0 + wm + mi * 16 + (lane >> 4) * 4 + reg;
        if (row < p.O) {
          if (nsplit == 1)
            dw[wbase + (long)row * Kw + col] = from_f32<T>(acc[mi][ni][reg]);
          else
            atomicAdd(&ws[wbase + (long)row * Kw + col], acc[mi][ni][reg]);
        }
      }
    }
}

// ---------------- launchers ----------------
void launch_conv2d_fwd_bf16(bf16* y, const bf16* x, const bf16* w,
                            const ConvParams& p, hipStream_t s) {
  long P = (long)p.OH * p.OW;
  if (P >= 4096) {
    dim3 grid(ceil_div(P, 128), ceil_div(p.O, 128), p.B);
    hipLaunchKernelGGL(conv2d_fwd_bf16_128, grid, dim3(256), 0, s, y, x, w,
                       p);
  } else {
    dim3 grid(ceil_div(P, 64), ceil_div(p.O, 64), p.B);
    hipLaunchKernelGGL(conv2d_fwd_bf16, grid, dim3(256), 0, s, y, x, w, p);
  }
}
void launch_conv2d_fwd_f32(float* y, const float* x, const float* w,
                           const ConvParams& p, hipStream_t s) {
  dim3 grid(ceil_div(p.OH * (long)p.OW, 64), ceil_div(p.O, 64), p.B);
  hipLaunchKernelGGL(conv2d_fwd_f32, grid, dim3(256), 0, s, y, x, w, p);
}
int conv2d_wgrad_nsplit(const ConvParams& p) {
  // choose split-K so the grid has >~1024 workgroups (256 CUs x 4)
  int Kw = p.I * p.kh * p.kw;
  long natural = (long)ceil_div(Kw, 64) * ceil_div(p.O, 64)
                 * (p.per_sample ? p.B : 1);
  int iters = ceil_div((long)p.OH * p.OW, 32) * (p.per_sample ? 1 : p.B);
  int nsplit = (int)((1024 + natural - 1) / natural);
  if (nsplit > iters) nsplit = iters;
  if (nsplit > 64) nsplit = 64;
  if (nsplit < 1) nsplit = 1;
  return nsplit;
}

template <typename T>
void launch_conv2d_wgrad(T* dw, float* ws, const T* x, const T* dy,
                         const ConvParams& p, int nsplit, hipStream_t s) {
  int Kw = p.I * p.kh * p.kw;
  dim3 grid(ceil_div(Kw, 64), ceil_div(p.O, 64),
            (p.per_sample ? p.B : 1) * nsplit);
  hipLaunchKernelGGL(conv2d_wgrad_kernel<T>, grid, dim3(256), 0, s, dw, ws,
                     x, dy, p, nsplit);
}
template void launch_conv2d_wgrad<float>(float*, float*, const float*,
                                         const float*, const ConvParams&,
                                         int, hipStream_t);
template void launch_conv2d_wgrad<bf16>(bf16*, float*, const bf16*,
                                        const bf16*, const ConvParams&, int,
                                        hipStream_t);

}  // namespace gfa
