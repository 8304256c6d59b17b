// Stride-2 3x3 conv (pad 1) on the LDS-slab pattern, for gfx950.
// Serves the discriminator's downsample convs and the input-gradient of
// the parity up-conv (dx of conv2d_up2 is a stride-2 conv of dY).
//
// Same structure as conv2d_slab.hip: tap-major K order, weights
// pre-permuted [O][9][I] read at use from L2, the input window staged in
// LDS per 32-channel block with a split load/write phase. Stride 2
// means the 16x8 output tile's window is 33x17 input pixels (+pad), so
// the slab is single-buffered (49 KB); the next block's loads are still
// issued before the tap loop and only the LDS write sits between the
// two barriers per channel block.
//
// Replaces the cuDNN strided convs of the TF reference.
#include "common.h"

namespace gfa {

namespace {
constexpr int S2_PIX = 40;                 // u16 per slab pixel (32+8)
constexpr int S2_W = 34;                   // slab cols  (33 + 1 pad)
constexpr int S2_H = 17;                   // slab rows
constexpr int S2_N = S2_H * S2_W * S2_PIX;
}  // namespace

__global__ __launch_bounds__(256, 2)
void conv2d_s2_slab_bf16(bf16* __restrict__ y, const bf16* __restrict__ x,
                         const bf16* __restrict__ wr,  // [O][9][I]
                         int B, int I, int H, int W, int O) {
  __shared__ u16 slab[S2_N];

  const int OH = H >> 1, OW = W >> 1;
  const int tilesX = OW >> 4;
  const int b = blockIdx.z;
  const int m0 = blockIdx.y * 128;
  const int ty = blockIdx.x / tilesX, tx = blockIdx.x % tilesX;
  const int y0 = ty * 8, x0 = tx * 16;     // OUTPUT-space tile origin
  const int ix0 = 2 * x0, iy0 = 2 * y0;    // input-space

  const bf16* xb = x + (long)b * I * H * W;
  bf16* yb = y + (long)b * O * OH * OW;

  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const int wm = (wave >> 1) * 64;
  const int wpy = (wave & 1) * 4;
  const int px = lane & 15;
  const int ag = lane >> 4;

  int a_off[4];
#pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
    const int o = min(m0 + wm + mi * 16 + (lane & 15), O - 1);
    a_off[mi] = o * 9 * I + ag * 8;
  }

  const int nCB = I >> 5;

  // 544 (dy 0..16, c) row tasks; window [ix0-4, ix0+36) as 10 s16x4.
  // Threads 0..31 carry a third task.
  const int n_task = (t < 544 - 512) ? 3 : 2;
  s16x4 tk[3][10];

  auto slab_load = [&](int c0) {
#pragma unroll
    for (int k = 0; k < 3; ++k) {
      if (k >= n_task) break;
      const int idx = t + k * 256;
      const int dy = idx >> 5, c = idx & 31;
      const int gy = iy0 + dy - 1;
      const bf16* src = xb + ((long)(c0 + c) * H + gy) * W;
      if (gy < 0 || gy >= H) {
#pragma unroll
        for (int j = 0; j < 10; ++j) tk[k][j] = s16x4{};
      } else if (ix0 >= 4 && ix0 + 36 <= W) {
        const s16x4* sp = reinterpret_cast<const s16x4*>(src + ix0 - 4);
#pragma unroll
        for (int j = 0; j < 10; ++j) tk[k][j] = sp[j];
      } else if (W == 32) {
        // single-tile rows: [0,32) direct into tk[1..8], zeros off ends
        const s16x4* sp = reinterpret_cast<const s16x4*>(src);
        tk[k][0] = s16x4{};
#pragma unroll
        for (int j = 0; j < 8; ++j) tk[k][1 + j] = sp[j];
        tk[k][9] = s16x4{};
      } else if (ix0 == 0) {
        // left edge: direct loads; tk[0]=0 keeps the rp[3+dx] mapping
        const s16x4* sp = reinterpret_cast<const s16x4*>(src);
        tk[k][0] = s16x4{};
#pragma unroll
        for (int j = 0; j < 9; ++j) tk[k][1 + j] = sp[j];
      } else {
        // right edge (ix0+32 == W): [W-36, W) direct; same rp offset
        const s16x4* sp = reinterpret_cast<const s16x4*>(src + W - 36);
#pragma unroll
        for (int j = 0; j < 9; ++j) tk[k][j] = sp[j];
        tk[k][9] = s16x4{};
      }
    }
  };

  auto slab_write = [&]() {
#pragma unroll
    for (int k = 0; k < 3; ++k) {
      if (k >= n_task) break;
      const int idx = t + k * 256;
      const int dy = idx >> 5, c = idx & 31;
      u16* dst = &slab[(dy * S2_W) * S2_PIX + c];
      const u16* rp = reinterpret_cast<const u16*>(&tk[k][0]);
#pragma unroll
      for (int dx = 0; dx < 34; ++dx) dst[dx * S2_PIX] = rp[3 + dx];
    }
  };

  f32x4 acc[4][4] = {};

  slab_load(0);
  slab_write();
  __syncthreads();

  for (int cb = 0; cb < nCB; ++cb) {
    const int c0 = cb << 5;
    const bool pre = cb + 1 < nCB;
    if (pre) slab_load(c0 + 32);
#pragma unroll
    for (int tap = 0; tap < 9; ++tap) {
      const int r = tap / 3, s = tap - r * 3;
      const int tc = tap * I + c0;
      s16x8 af[4];
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
        af[mi] = *reinterpret_cast<const s16x8*>(wr + a_off[mi] + tc);
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        const int py = wpy + ni;
        // input row 2(y0+py)+r-1 -> slab row 2py+r; col 2px+s
        const s16x8 bfr = *reinterpret_cast<const s16x8*>(
            &slab[((2 * py + r) * S2_W + 2 * px + s) * S2_PIX + (ag << 3)]);
#pragma unroll
        for (int mi = 0; mi < 4; ++mi)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mi], bfr, acc[mi][ni], 0, 0, 0);
      }
    }
    __syncthreads();           // tap-loop reads of the slab are done
    if (pre) {
      slab_write();
      __syncthreads();         // new slab visible for the next block
    }
  }

#pragma unroll
  for (int ni = 0; ni < 4; ++ni) {
    const int gy = y0 + wpy + ni;
    const int gx = x0 + px;
    bf16* yp = yb + (long)gy * OW + gx;
#pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
      const int o0 = m0 + wm + mi * 16 + ag * 4;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int o = o0 + reg;
        if (o < O) yp[(long)o * OH * OW] = from_f32<bf16>(acc[mi][ni][reg]);
      }
    }
  }
}

bool conv2d_s2_eligible(int I, int O, int H, int W, int OH, int OW, int kh,
                        int kw, int stride, int pad, int per_sample) {
  return !per_sample && kh == 3 && kw == 3 && stride == 2 && pad == 1 &&
         (I & 31) == 0 && I >= 32 && (H & 15) == 0 && (W & 31) == 0 &&
         W >= 32 && OH == H / 2 && OW == W / 2;
}

void launch_conv2d_s2_slab_bf16(bf16* y, const bf16* x, const bf16* wr,
                                int B, int I, int H, int W, int O,
                                hipStream_t s) {
  dim3 grid((W >> 5) * (H >> 4), ceil_div(O, 128), B);
  hipLaunchKernelGGL(conv2d_s2_slab_bf16, grid, dim3(256), 0, s, y, x, wr,
                     B, I, H, W, O);
}

}  // namespace gfa
