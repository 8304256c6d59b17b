// Parity-decomposed 2x-upsampling 3x3 conv (transposed conv) for gfx950.
//
//   y = conv2d(zero_stuff2(x), w, pad=1)        y: [B,O,2H,2W]
//
// The zero-stuffed conv touches a real input element only when the tap
// parity matches the output parity, so each of the 4 output parity
// classes (oy&1, ox&1) is a small dense conv AT INPUT RESOLUTION with
// 1/2/2/4 of the 9 taps:
//
//   ey=0: r=1          ey=1: r in {0,2}   (x-row offset d=(ey+r-1)/2)
//   ex=0: s=1          ex=1: s in {0,2}
//
// Total MACs = O*I*9*H*W — 4x fewer than blur-upsample-then-conv at
// output resolution, which is what the TF reference's graph did on
// cuDNN (upsample_conv_2d). The G synthesis up-convs
// are the largest convs in the model, so this is a 4x on ~half of G.
//
// Same LDS slab staging as conv2d_slab.hip (16x8 input tile + halo per
// 32-channel block; split load/write phases; at-use A fragments from
// L2-hot [O][9][I] weights). Tile: 64(O) x 32x16 output pixels
// (= 16x8 input pixels x 4 parities); per wave 2(M)x4(N)x4(parity)
// accumulators, 72 MFMA per channel block per wave.
#include "common.h"

namespace gfa {

namespace {
constexpr int SLAB_PIX = 40;
constexpr int SLAB_N = 10 * 18 * SLAB_PIX;

// parity -> (weight tap index r*3+s, slab row offset, slab col offset);
// constexpr so the tap loops fully unroll with immediate offsets
struct Tap { int w, dr, ds; };
constexpr Tap kTaps[4][4] = {
    /* ey0 ex0 */ {{4, 0, 0}, {0, 0, 0}, {0, 0, 0}, {0, 0, 0}},
    /* ey0 ex1 */ {{3, 0, 0}, {5, 0, 1}, {0, 0, 0}, {0, 0, 0}},
    /* ey1 ex0 */ {{1, 0, 0}, {7, 1, 0}, {0, 0, 0}, {0, 0, 0}},
    /* ey1 ex1 */ {{0, 0, 0}, {2, 0, 1}, {6, 1, 0}, {8, 1, 1}},
};
constexpr int kNTaps[4] = {1, 2, 2, 4};
}  // namespace

__global__ __launch_bounds__(256, 2)
void conv2d_up2_slab_bf16(bf16* __restrict__ y, const bf16* __restrict__ x,
                          const bf16* __restrict__ wr,  // [O][9][I]
                          int B, int I, int H, int W, int O) {
  __shared__ u16 slab[2][SLAB_N];

  const int tilesX = W >> 4;
  const int b = blockIdx.z;
  const int m0 = blockIdx.y * 64;
  const int ty = blockIdx.x / tilesX, tx = blockIdx.x % tilesX;
  const int y0 = ty * 8, x0 = tx * 16;           // input-space tile origin

  const bf16* xb = x + (long)b * I * H * W;
  const int OH = 2 * H, OW = 2 * W;
  bf16* yb = y + (long)b * O * OH * OW;

  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const int wm = (wave >> 1) * 32;               // O offset of this wave
  const int wpy = (wave & 1) * 4;                // input-row offset
  const int px = lane & 15;
  const int ag = lane >> 4;

  int a_off[2];
#pragma unroll
  for (int mi = 0; mi < 2; ++mi) {
    const int o = min(m0 + wm + mi * 16 + (lane & 15), O - 1);
    a_off[mi] = o * 9 * I + ag * 8;
  }

  const int nCB = I >> 5;

  const int n_task = (t < 320 - 256) ? 2 : 1;
  s16x4 tk[2][6];

  auto slab_load = [&](int c0) {
#pragma unroll
    for (int k = 0; k < 2; ++k) {
      if (k >= n_task) break;
      const int idx = t + k * 256;
      const int dy = idx >> 5, c = idx & 31;
      const int gy = y0 + dy - 1;
      const bf16* src = xb + ((long)(c0 + c) * H + gy) * W;
      if (gy < 0 || gy >= H) {
#pragma unroll
        for (int j = 0; j < 6; ++j) tk[k][j] = s16x4{};
      } else if (x0 >= 4 && x0 + 20 <= W) {
        const s16x4* sp = reinterpret_cast<const s16x4*>(src + x0 - 4);
#pragma unroll
        for (int j = 0; j < 6; ++j) tk[k][j] = sp[j];
      } else if (x0 == 0 && W >= 24) {
        // left edge: load [0,20) DIRECTLY into tk[1..5]; tk[0]=0 keeps
        // the rp[3+dx] <-> x[x0-1+dx] mapping (register repacking after
        // the loads would force a vmcnt(0) wait before the MFMA stream)
        const s16x4* sp = reinterpret_cast<const s16x4*>(src);
        tk[k][0] = s16x4{};
#pragma unroll
        for (int j = 0; j < 5; ++j) tk[k][1 + j] = sp[j];
      } else if (x0 + 16 == W && W >= 24) {
        // right edge: load [W-24, W) directly; the write phase switches
        // to offset 7 (rp[i] <-> x[W-24+i]) and zeroes dx==17 (x==W)
        const s16x4* sp = reinterpret_cast<const s16x4*>(src + W - 24);
#pragma unroll
        for (int j = 0; j < 6; ++j) tk[k][j] = sp[j];
      } else {
        u16* rp = reinterpret_cast<u16*>(&tk[k][0]);
#pragma unroll
        for (int dx = 0; dx < 24; ++dx) {
          const int gx = x0 - 4 + dx;
          rp[dx] = (gx >= 0 && gx < W) ? __builtin_bit_cast(u16, src[gx])
                                       : (u16)0;
        }
      }
    }
  };

  const bool r_edge24 = (x0 + 16 == W) && (x0 >= 4) && (W >= 24);

  auto slab_write = [&](int sb) {
#pragma unroll
    for (int k = 0; k < 2; ++k) {
      if (k >= n_task) break;
      const int idx = t + k * 256;
      const int dy = idx >> 5, c = idx & 31;
      u16* dst = &slab[sb][(dy * 18) * SLAB_PIX + c];
      const u16* rp = reinterpret_cast<const u16*>(&tk[k][0]);
      if (r_edge24) {
#pragma unroll
        for (int dx = 0; dx < 17; ++dx) dst[dx * SLAB_PIX] = rp[7 + dx];
        dst[17 * SLAB_PIX] = 0;  // x == W
      } else {
#pragma unroll
        for (int dx = 0; dx < 18; ++dx) dst[dx * SLAB_PIX] = rp[3 + dx];
      }
    }
  };

  f32x4 acc[4][2][4] = {};  // [parity][mi][ni]

  slab_load(0);
  slab_write(0);
  __syncthreads();

  for (int cb = 0; cb < nCB; ++cb) {
    const int c0 = cb << 5;
    const int sb = cb & 1;
    const bool pre = cb + 1 < nCB;
    if (pre) slab_load(c0 + 32);
#pragma unroll
    for (int par = 0; par < 4; ++par) {
#pragma unroll
      for (int ti = 0; ti < 4; ++ti) {
        if (ti >= kNTaps[par]) break;
        const Tap tp = kTaps[par][ti];
        const int tc = tp.w * I + c0;
        s16x8 af[2];
#pragma unroll
        for (int mi = 0; mi < 2; ++mi)
          af[mi] = *reinterpret_cast<const s16x8*>(wr + a_off[mi] + tc);
#pragma unroll
        for (int ni = 0; ni < 4; ++ni) {
          const int py = wpy + ni;
          // slab row (py + dr) maps input row y0+py+dr-... halo base +1
          // wait: slab row index j holds input row y0 + j - 1, so input
          // row (y0 + py + dr) lives at slab row (py + dr + 1).
          const s16x8 bfr = *reinterpret_cast<const s16x8*>(
              &slab[sb][((py + tp.dr + 1) * 18 + px + tp.ds + 1) * SLAB_PIX +
                        (ag << 3)]);
#pragma unroll
          for (int mi = 0; mi < 2; ++mi)
            acc[par][mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                af[mi], bfr, acc[par][mi][ni], 0, 0, 0);
        }
      }
    }
    if (pre) slab_write(sb ^ 1);
    __syncthreads();
  }

  // epilogue: parity-interleaved stores into the 2H x 2W output
#pragma unroll
  for (int par = 0; par < 4; ++par) {
    const int ey = par >> 1, ex = par & 1;
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      const int gy = 2 * (y0 + wpy + ni) + ey;
      const int gx = 2 * (x0 + px) + ex;
      bf16* yp = yb + (long)gy * OW + gx;
#pragma unroll
      for (int mi = 0; mi < 2; ++mi) {
        const int o0 = m0 + wm + mi * 16 + ag * 4;
#pragma unroll
        for (int reg = 0; reg < 4; ++reg) {
          const int o = o0 + reg;
          if (o < O)
            yp[(long)o * OH * OW] = from_f32<bf16>(acc[par][mi][ni][reg]);
        }
      }
    }
  }
}

bool conv2d_up2_eligible(int I, int O, int H, int W, int kh, int kw,
                         int per_sample) {
  return !per_sample && kh == 3 && kw == 3 && (I & 31) == 0 && I >= 32 &&
         (H & 7) == 0 && (W & 15) == 0 && H >= 8 && W >= 16;
}

void launch_conv2d_up2_slab_bf16(bf16* y, const bf16* x, const bf16* wr,
                                 int B, int I, int H, int W, int O,
                                 hipStream_t s) {
  dim3 grid((W >> 4) * (H >> 3), ceil_div(O, 64), B);
  hipLaunchKernelGGL(conv2d_up2_slab_bf16, grid, dim3(256), 0, s, y, x, wr,
                     B, I, H, W, O);
}

}  // namespace gfa
