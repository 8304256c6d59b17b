// Tap-major LDS-slab implicit-GEMM 3x3 conv for gfx950 (stride 1, pad 1).
//
// The generic kernel in conv2d.hip decodes 9 filter taps per K-step and
// re-reads every input pixel up to 9 times from global memory; its
// MFMA:staging ratio makes it staging-bound (~9% of bf16 MFMA peak,
// profiles/r01_summary.md). This kernel reorders K as (tap, channel):
// the 16x16-pixel output tile's input slab (18x18xBC, halo included) is
// staged in LDS ONCE per 32-channel block, and the 9 taps' B-operand
// fragments are read DIRECTLY from the slab at shifted offsets — no
// per-tap global traffic, no per-tap LDS re-staging.
//
//   C[o][p]  +=  sum_{t,c} W[o][t][c] * slab[(py+ty(t))(px+tx(t))][c]
//
// Weights are passed pre-permuted as [O][9][I] so A-fragments are
// contiguous 16-B reads. A is double-buffered per tap; the slab for the
// next channel block is prefetched during tap 0 (its latency hides
// behind ~8 taps of MFMA).
//
// Tile: 128(O) x 128(pixels as 16 wide x 8 high), BC=32, 4 waves, each
// wave a 64x64 sub-tile = 4x4 fragments of 16x16,
// v_mfma_f32_16x16x32_bf16. Per wave per (channel-block, tap):
// 8 ds_read_b128 + 16 MFMA, at 3 waves/SIMD occupancy.
//
// Replaces the reference's cuDNN 3x3 convs (TF 1.14).
#include "common.h"

namespace gfa {

namespace {
constexpr int SLAB_PIX = 40;           // u16 per slab pixel (32 + 8 pad)
constexpr int SLAB_N = 10 * 18 * SLAB_PIX;  // one slab buffer, u16
constexpr int AROW = 40;               // u16 per A row (32 + 8 pad)
}  // namespace

__global__ __launch_bounds__(256, 3)
void conv2d_fwd_slab_bf16(bf16* __restrict__ y, const bf16* __restrict__ x,
                          const bf16* __restrict__ wr,  // [O][9][I]
                          int B, int I, int H, int W, int O) {
  __shared__ u16 slab[2][SLAB_N];
  __shared__ u16 As[2][128 * AROW];

  const int tilesX = W >> 4;
  const int b = blockIdx.z;
  const int m0 = blockIdx.y * 128;
  const int ty = blockIdx.x / tilesX, tx = blockIdx.x % tilesX;
  const int y0 = ty * 8, x0 = tx * 16;

  const bf16* xb = x + (long)b * I * H * W;
  bf16* yb = y + (long)b * O * H * W;

  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const int wm = (wave >> 1) * 64;        // M offset of this wave
  const int wpy = (wave & 1) * 4;         // tile-row offset of this wave
  const int px = lane & 15;               // B-fragment pixel x (fixed)
  const int ag = lane >> 4;               // K granule 0..3

  // ---- staging roles ----
  // slab: tasks (dy, c): idx -> dy = idx/32 in 0..17, c = idx%32.
  // A: thread -> row t>>1 (0..127), half g2 = t&1 (16 u16 each).
  const int a_row = t >> 1, a_g2 = t & 1;
  const long wrow = (long)(m0 + a_row) * 9 * I + a_g2 * 16;
  const bool a_ok = (m0 + a_row) < O;

  const int nCB = I >> 5;

  // part < 0: whole slab; part 0..2: one 256-task slice (spread across
  // taps so each tap segment's barrier only waits on 1/3 of the loads)
  auto stage_slab = [&](int c0, int sb, int part) {
    const int lo = part < 0 ? 0 : part * 160;
    const int hi = part < 0 ? 10 * 32 : min(10 * 32, lo + 160);
    for (int idx = lo + t; idx < hi; idx += 256) {
      const int dy = idx >> 5, c = idx & 31;
      const int gy = y0 + dy - 1;
      u16* dst = &slab[sb][(dy * 18) * SLAB_PIX + c];
      const bf16* src = xb + ((long)(c0 + c) * H + gy) * W;
      if (gy < 0 || gy >= H) {
#pragma unroll
        for (int dx = 0; dx < 18; ++dx) dst[dx * SLAB_PIX] = 0;
      } else if (x0 >= 4 && x0 + 20 <= W) {
        // interior: 6 aligned 8-B loads covering [x0-4, x0+20)
        s16x4 r[6];
        const s16x4* sp = reinterpret_cast<const s16x4*>(src + x0 - 4);
#pragma unroll
        for (int j = 0; j < 6; ++j) r[j] = sp[j];
        const u16* rp = reinterpret_cast<const u16*>(&r[0]);
#pragma unroll
        for (int dx = 0; dx < 18; ++dx) dst[dx * SLAB_PIX] = rp[3 + dx];
      } else {
#pragma unroll
        for (int dx = 0; dx < 18; ++dx) {
          const int gx = x0 - 1 + dx;
          u16 v = 0;
          if (gx >= 0 && gx < W)
            v = __builtin_bit_cast(u16, src[gx]);
          dst[dx * SLAB_PIX] = v;
        }
      }
    }
  };

  auto stage_A = [&](int tap, int c0, int ab) {
    u16* dst = &As[ab][a_row * AROW + a_g2 * 16];
    if (a_ok) {
      const s16x8* sp =
          reinterpret_cast<const s16x8*>(wr + wrow + (long)tap * I + c0);
      *reinterpret_cast<s16x8*>(dst) = sp[0];
      *reinterpret_cast<s16x8*>(dst + 8) = sp[1];
    } else {
      *reinterpret_cast<s16x8*>(dst) = s16x8{};
      *reinterpret_cast<s16x8*>(dst + 8) = s16x8{};
    }
  };

  f32x4 acc[4][4] = {};

  stage_slab(0, 0, -1);
  stage_A(0, 0, 0);
  __syncthreads();

  for (int cb = 0; cb < nCB; ++cb) {
    const int c0 = cb << 5;
    const int sb = cb & 1;
#pragma unroll
    for (int tap = 0; tap < 9; ++tap) {
      // prefetch: next A buffer; during tap 0, also next channel slab
      if (tap < 8) {
        stage_A(tap + 1, c0, (tap + 1) & 1);
      } else if (cb + 1 < nCB) {
        stage_A(0, c0 + 32, 1);  // tap 9 ≡ buffer (9)&1 = 1
      }
      if (tap < 2 && cb + 1 < nCB) stage_slab(c0 + 32, sb ^ 1, tap);

      const int r = tap / 3, s = tap - r * 3;
      const u16* ab = &As[tap & 1][0];
      // A fragments for this wave (4), B fragments (8), 32 MFMA
      s16x8 afr[4];
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
        afr[mi] = *reinterpret_cast<const s16x8*>(
            &ab[(wm + mi * 16 + (lane & 15)) * AROW + (ag << 3)]);
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        const int py = wpy + ni;
        const s16x8 bfr = *reinterpret_cast<const s16x8*>(
            &slab[sb][((py + r) * 18 + px + s) * SLAB_PIX + (ag << 3)]);
#pragma unroll
        for (int mi = 0; mi < 4; ++mi)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afr[mi], bfr, acc[mi][ni], 0, 0, 0);
      }
      __syncthreads();
    }
  }

  // ---- epilogue: scalar bf16 stores (stride H*W between o rows) ----
#pragma unroll
  for (int ni = 0; ni < 4; ++ni) {
    const int gy = y0 + wpy + ni;
    const int gx = x0 + px;
    bf16* yp = yb + (long)gy * W + gx;
#pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
      const int o0 = m0 + wm + mi * 16 + ag * 4;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int o = o0 + reg;
        if (o < O) yp[(long)o * H * W] = from_f32<bf16>(acc[mi][ni][reg]);
      }
    }
  }
}

bool conv2d_slab_eligible(int I, int O, int H, int W, int OH, int OW,
                          int kh, int kw, int stride, int pad,
                          int per_sample) {
  return !per_sample && kh == 3 && kw == 3 && stride == 1 && pad == 1 &&
         (I & 31) == 0 && I >= 32 && (H & 7) == 0 && (W & 15) == 0 &&
         H >= 8 && W >= 16 && OH == H && OW == W;
}

void launch_conv2d_fwd_slab_bf16(bf16* y, const bf16* x, const bf16* wr,
                                 int B, int I, int H, int W, int O,
                                 hipStream_t s) {
  dim3 grid((W >> 4) * (H >> 3), ceil_div(O, 128), B);
  hipLaunchKernelGGL(conv2d_fwd_slab_bf16, grid, dim3(256), 0, s, y, x, wr,
                     B, I, H, W, O);
}

}  // namespace gfa
