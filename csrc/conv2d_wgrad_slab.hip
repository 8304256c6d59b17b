// Tap-major weight-gradient kernel for 3x3 stride-1 convs on gfx950.
//
// dw[o][r][s][c] = sum_{b,p} dy[b][o][p] * x[b][c][p + (r-1, s-1)]
//
// The generic wgrad in conv2d.hip re-stages dY and x patches per 64x64
// output tile with 2x2 fragments/wave — staging-bound. Here:
//
//  * K order is pixels, in 32-wide row chunks; each k-step's x window
//    (32 ch x 3 rows x 34 cols) is loaded from global ONCE and expanded
//    in LDS into its 9 tap-shifted copies (an im2col image in LDS: the
//    9x read amplification happens on LDS bandwidth, not HBM), so every
//    B fragment is an aligned 16-B ds_read.
//  * dY (the A operand) is staged through LDS once per k-step (16-B
//    chunk tasks over all 256 threads): the per-fragment alternative has
//    16 lanes hitting 16 strided rows per instruction and parks the
//    wave. Blocks are additionally swizzled so all I/32 channel-tiles of
//    one (O-tile, K-split) land on the SAME XCD (blockIdx%8 picks the
//    XCD) and share the dY slice through that XCD's L2.
//  * One block computes dw for 128 O x 32 C x all 9 taps; per wave
//    4(M) x 1(C) x 9(tap) accumulators, 36 MFMA per 32-pixel k-step
//    against 4 A-fragment loads. f32 atomics accumulate across K-splits.
//
// Replaces the cuDNN/TF wgrad of the reference.
#include "common.h"

namespace gfa {

namespace {
constexpr int SLABX = 40;               // u16 per (c, r, s) pixel row
constexpr int SLAB_N = 32 * 3 * 3 * SLABX;  // one buffer: c, r, s, px
}  // namespace

__global__ __launch_bounds__(256, 2)
void conv2d_wgrad_slab_bf16(float* __restrict__ ws,  // [O][9][I] f32
                            const bf16* __restrict__ x,
                            const bf16* __restrict__ dy,
                            int B, int I, int H, int W, int O, int nsplit) {
  __shared__ u16 slab[2][SLAB_N];
  __shared__ u16 ald[2][128 * 40];  // dY tile [row][32px + 8 pad]

  // block swizzle: bx = xcd + 8*(ct + nCt*pairHi); pair = xcd + 8*pairHi
  const int nCt = I >> 5;
  const int bx = blockIdx.x;
  const int xcd = bx & 7;
  const int rest = bx >> 3;
  const int ct = rest % nCt;
  const int pair = xcd + 8 * (rest / nCt);
  const int nMt = (O + 127) >> 7;
  const int m0 = (pair % nMt) << 7;
  const int split = pair / nMt;
  const int c0 = ct << 5;
  if (pair >= nMt * nsplit) return;  // swizzle padding

  const int OWc = W >> 5;                     // 32-px chunks per row
  const long iters = (long)B * H * OWc;
  const long span = (iters + nsplit - 1) / nsplit;
  const long it0 = (long)split * span;
  const long it1 = min(iters, it0 + span);
  if (it0 >= it1) return;

  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const int wm = (wave >> 1) * 64;            // O offset of this wave
  const int wc = (wave & 1) * 16;             // C offset of this wave
  const int ag = lane >> 4;                   // pixel granule 0..3

  // dY staging: 512 (row, granule) 16-B chunk tasks over 256 threads
  // (a direct per-fragment read had 16 lanes hitting 16 strided rows:
  // a dY-constant experiment ran 1.43x faster)
  const int dy_lrow = t >> 2;                 // tile rows r and r+64
  const int dy_row = min(m0 + dy_lrow, O - 1);
  const int dy_row2 = min(m0 + 64 + dy_lrow, O - 1);
  const int dy_g = t & 3;                     // 16-B granule 0..3
  s16x8 dk[2];

  // staging: 384 (c, dr, quarter) tasks of 16-element x windows spread
  // over ALL 256 threads (a 96-full-row split left waves 2-3 idle and
  // every barrier waiting on wave 0). Task (c, dr, q) covers dst pixels
  // [8q, 8q+8) of row (c, dr): window [col0-4+8q, +16) as 4 s16x4.
  const int n_task = (t < 384 - 256) ? 2 : 1;
  s16x4 tk[2][4];

  auto win_load = [&](long it) {
    const int b = (int)(it / ((long)H * OWc));
    const int rem = (int)(it - (long)b * H * OWc);
    const int row = rem / OWc;
    const int col0 = (rem - row * OWc) << 5;
#pragma unroll
    for (int k = 0; k < 2; ++k) {
      if (k >= n_task) break;
      const int idx = t + k * 256;
      const int c = idx & 31, rest = idx >> 5;
      const int dr = rest % 3, q = rest / 3;
      const int gy = row + dr - 1;
      const int lo = col0 - 4 + 8 * q;
      const bf16* src = x + (((long)b * I + c0 + c) * H + gy) * W;
      if (gy < 0 || gy >= H) {
#pragma unroll
        for (int j = 0; j < 4; ++j) tk[k][j] = s16x4{};
      } else if (lo >= 0 && lo + 16 <= W) {
        const s16x4* sp = reinterpret_cast<const s16x4*>(src + lo);
#pragma unroll
        for (int j = 0; j < 4; ++j) tk[k][j] = sp[j];
      } else if (lo < 0) {
        // left edge: lo == -4 exactly; x<0 slots are zero
        const s16x4* sp = reinterpret_cast<const s16x4*>(src);
        tk[k][0] = s16x4{};
#pragma unroll
        for (int j = 0; j < 3; ++j) tk[k][1 + j] = sp[j];
      } else {
        // right edge: lo == W-12 exactly; x>=W slots are zero
        const s16x4* sp = reinterpret_cast<const s16x4*>(src + W - 16);
#pragma unroll
        for (int j = 0; j < 3; ++j) tk[k][j] = sp[1 + j];
        tk[k][3] = s16x4{};
      }
    }
  };

  // expand each register window into its slice of the 3 tap-shifted
  // LDS copies: dst[s][8q + jj] = window[s + 3 + jj]
  auto win_write = [&](int sb) {
#pragma unroll
    for (int k = 0; k < 2; ++k) {
      if (k >= n_task) break;
      const int idx = t + k * 256;
      const int c = idx & 31, rest = idx >> 5;
      const int dr = rest % 3, q = rest / 3;
      const u16* rp = reinterpret_cast<const u16*>(&tk[k][0]);
      u16* base = &slab[sb][(c * 3 + dr) * 3 * SLABX + 8 * q];
#pragma unroll
      for (int s = 0; s < 3; ++s) {
        u16* dst = base + s * SLABX;
#pragma unroll
        for (int jj = 0; jj < 8; ++jj) dst[jj] = rp[s + 3 + jj];
      }
    }
  };

  f32x4 acc[9][4] = {};

  win_load(it0);
  win_write(0);

  auto dy_load = [&](long it) {
    const int b = (int)(it / ((long)H * OWc));
    const int rem = (int)(it - (long)b * H * OWc);
    const long dybase = ((long)b * O) * H * W + (long)rem * 32 + dy_g * 8;
    dk[0] = *reinterpret_cast<const s16x8*>(dy + dybase +
                                            (long)dy_row * H * W);
    dk[1] = *reinterpret_cast<const s16x8*>(dy + dybase +
                                            (long)dy_row2 * H * W);
  };
  auto dy_write = [&](int sb) {
    *reinterpret_cast<s16x8*>(&ald[sb][dy_lrow * 40 + dy_g * 8]) = dk[0];
    *reinterpret_cast<s16x8*>(&ald[sb][(64 + dy_lrow) * 40 + dy_g * 8]) =
        dk[1];
  };

  dy_load(it0);
  dy_write(0);
  __syncthreads();  // note: pairs with the prologue win_write barrier

  for (long it = it0; it < it1; ++it) {
    const int sb = (int)((it - it0) & 1);
    if (it + 1 < it1) {
      win_load(it + 1);
      dy_load(it + 1);
    }

    const u16* al = &ald[sb][0];
    s16x8 af[4];
#pragma unroll
    for (int mi = 0; mi < 4; ++mi)
      af[mi] = *reinterpret_cast<const s16x8*>(
          &al[(wm + mi * 16 + (lane & 15)) * 40 + (ag << 3)]);

    const u16* sl = &slab[sb][0];
#pragma unroll
    for (int tap = 0; tap < 9; ++tap) {
      const int r = tap / 3, s = tap - r * 3;
      // B fragment: 16 channels (lane&15) x 8 pixels (granule)
      const s16x8 bfr = *reinterpret_cast<const s16x8*>(
          &sl[(((wc + (lane & 15)) * 3 + r) * 3 + s) * SLABX + (ag << 3)]);
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
        acc[tap][mi] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            af[mi], bfr, acc[tap][mi], 0, 0, 0);
    }
    if (it + 1 < it1) {
      win_write(sb ^ 1);
      dy_write(sb ^ 1);
    }
    __syncthreads();
  }

  // epilogue: C rows = O, cols = channels; atomic f32 accumulate
#pragma unroll
  for (int tap = 0; tap < 9; ++tap) {
    const int c = c0 + wc + (lane & 15);
#pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
      const int o0 = m0 + wm + mi * 16 + ag * 4;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int o = o0 + reg;
        if (o < O)
          atomicAdd(&ws[((long)o * 9 + tap) * I + c], acc[tap][mi][reg]);
      }
    }
  }
}

// ---- stride-2 variant ------------------------------------------------
// Same tap-major structure; a k-step is 32 OUTPUT pixels of one dY row,
// whose x window is 66 input columns x 3 rows. Each (c, dr) row splits
// into two overlapping 40-element half-windows (threads 0..191, one
// each) so register staging stays at 10 s16x4 per thread; the s-shifted
// LDS copies subsample the window at stride 2.
__global__ __launch_bounds__(256, 2)
void conv2d_wgrad_slab_s2_bf16(float* __restrict__ ws,  // [O][9][I] f32
                               const bf16* __restrict__ x,
                               const bf16* __restrict__ dy,
                               int B, int I, int H, int W, int O,
                               int nsplit) {
  __shared__ u16 slab[2][SLAB_N];
  __shared__ u16 ald[2][128 * 40];  // dY tile, staged like the s1 kernel

  const int OH = H >> 1, OW = W >> 1;
  const int nCt = I >> 5;
  const int bx = blockIdx.x;
  const int xcd = bx & 7;
  const int rest = bx >> 3;
  const int ct = rest % nCt;
  const int pair = xcd + 8 * (rest / nCt);
  const int nMt = (O + 127) >> 7;
  const int m0 = (pair % nMt) << 7;
  const int split = pair / nMt;
  const int c0 = ct << 5;
  if (pair >= nMt * nsplit) return;

  const int OWc = OW >> 5;
  const long iters = (long)B * OH * OWc;
  const long span = (iters + nsplit - 1) / nsplit;
  const long it0 = (long)split * span;
  const long it1 = min(iters, it0 + span);
  if (it0 >= it1) return;

  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const int wm = (wave >> 1) * 64;
  const int wc = (wave & 1) * 16;
  const int ag = lane >> 4;

  const int dy_lrow = t >> 2;
  const int dy_row = min(m0 + dy_lrow, O - 1);
  const int dy_row2 = min(m0 + 64 + dy_lrow, O - 1);
  const int dy_g = t & 3;
  s16x8 dk[2];

  // 192 (c, dr, half) tasks; half-window h of 40 elems at 2*col0-4+32h
  const int s_c = t & 31, s_dr = (t >> 5) % 3, s_h = t >> 5 >= 3;
  const bool stager = t < 192;
  s16x4 tk[10];

  auto win_load = [&](long it) {
    if (!stager) return;
    const int b = (int)(it / ((long)OH * OWc));
    const int rem = (int)(it - (long)b * OH * OWc);
    const int row = rem / OWc;
    const int col0 = (rem - row * OWc) << 5;
    const int gy = 2 * row + s_dr - 1;
    const int lo = 2 * col0 - 4 + 32 * s_h;
    const bf16* src = x + (((long)b * I + c0 + s_c) * H + gy) * W;
    if (gy < 0 || gy >= H) {
#pragma unroll
      for (int j = 0; j < 10; ++j) tk[j] = s16x4{};
    } else if (lo >= 0 && lo + 40 <= W) {
      const s16x4* sp = reinterpret_cast<const s16x4*>(src + lo);
#pragma unroll
      for (int j = 0; j < 10; ++j) tk[j] = sp[j];
    } else if (lo < 0) {
      // left edge: lo is always -4 (col0==0, h==0); rp[0..3] are x<0
      const s16x4* sp = reinterpret_cast<const s16x4*>(src);
      s16x4 tmp[9];
#pragma unroll
      for (int j = 0; j < 9; ++j) tmp[j] = sp[j];
      tk[0] = s16x4{};
#pragma unroll
      for (int j = 0; j < 9; ++j) tk[1 + j] = tmp[j];
    } else {
      // right edge: lo - (W-40) is always +4; rp[36..39] are x>=W
      const s16x4* sp = reinterpret_cast<const s16x4*>(src + W - 40);
      s16x4 tmp[9];
#pragma unroll
      for (int j = 0; j < 9; ++j) tmp[j] = sp[1 + j];
#pragma unroll
      for (int j = 0; j < 9; ++j) tk[j] = tmp[j];
      tk[9] = s16x4{};
    }
  };

  // copy s, dst j: x column 2*(col0+j)+s-1 = window[2j+s+3-32h], and
  // with j0 = 16*s_h the index 2*(j0+j)+s+3-32*s_h folds to 2j+s+3 —
  // written in the folded form so every tk access is a compile-time
  // constant and tk stays in registers (the unfolded runtime-s_h form
  // forced tk to 96 B/lane of scratch; see KNOWN_ISSUES on the
  // batch-64 scratch-fault suspect)
  auto win_write = [&](int sb) {
    if (!stager) return;
    const u16* rp = reinterpret_cast<const u16*>(&tk[0]);
    u16* base = &slab[sb][(s_c * 3 + s_dr) * 3 * SLABX];
    const int j0 = 16 * s_h;
#pragma unroll
    for (int s = 0; s < 3; ++s) {
      u16* dst = base + s * SLABX;
#pragma unroll
      for (int j = 0; j < 16; ++j)
        dst[j0 + j] = rp[2 * j + s + 3];
    }
  };

  f32x4 acc[9][4] = {};

  auto dy_load = [&](long it) {
    const int b = (int)(it / ((long)OH * OWc));
    const int rem = (int)(it - (long)b * OH * OWc);
    const long dybase = ((long)b * O) * OH * OW + (long)rem * 32 + dy_g * 8;
    dk[0] = *reinterpret_cast<const s16x8*>(dy + dybase +
                                            (long)dy_row * OH * OW);
    dk[1] = *reinterpret_cast<const s16x8*>(dy + dybase +
                                            (long)dy_row2 * OH * OW);
  };
  auto dy_write = [&](int sb) {
    *reinterpret_cast<s16x8*>(&ald[sb][dy_lrow * 40 + dy_g * 8]) = dk[0];
    *reinterpret_cast<s16x8*>(&ald[sb][(64 + dy_lrow) * 40 + dy_g * 8]) =
        dk[1];
  };

  win_load(it0);
  win_write(0);
  dy_load(it0);
  dy_write(0);
  __syncthreads();

  for (long it = it0; it < it1; ++it) {
    const int sb = (int)((it - it0) & 1);
    if (it + 1 < it1) {
      win_load(it + 1);
      dy_load(it + 1);
    }

    const u16* al = &ald[sb][0];
    s16x8 af[4];
#pragma unroll
    for (int mi = 0; mi < 4; ++mi)
      af[mi] = *reinterpret_cast<const s16x8*>(
          &al[(wm + mi * 16 + (lane & 15)) * 40 + (ag << 3)]);

    const u16* sl = &slab[sb][0];
#pragma unroll
    for (int tap = 0; tap < 9; ++tap) {
      const int r = tap / 3, s = tap - r * 3;
      const s16x8 bfr = *reinterpret_cast<const s16x8*>(
          &sl[(((wc + (lane & 15)) * 3 + r) * 3 + s) * SLABX + (ag << 3)]);
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
        acc[tap][mi] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            af[mi], bfr, acc[tap][mi], 0, 0, 0);
    }
    if (it + 1 < it1) {
      win_write(sb ^ 1);
      dy_write(sb ^ 1);
    }
    __syncthreads();
  }

#pragma unroll
  for (int tap = 0; tap < 9; ++tap) {
    const int c = c0 + wc + (lane & 15);
#pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
      const int o0 = m0 + wm + mi * 16 + ag * 4;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int o = o0 + reg;
        if (o < O)
          atomicAdd(&ws[((long)o * 9 + tap) * I + c], acc[tap][mi][reg]);
      }
    }
  }
}

bool conv2d_wgrad_slab_eligible(int I, int O, int H, int W, int OH, int OW,
                                int kh, int kw, int stride, int pad,
                                int per_sample) {
  if (per_sample || kh != 3 || kw != 3 || pad != 1 || (I & 31) != 0 ||
      I < 32)
    return false;
  if (stride == 1)
    return (W & 31) == 0 && W >= 32 && OH == H && OW == W;
  if (stride == 2)
    return (W & 63) == 0 && W >= 64 && (H & 1) == 0 && OH == H / 2 &&
           OW == W / 2;
  return false;
}

int conv2d_wgrad_slab_nsplit(int B, int I, int H, int W, int O, int stride) {
  const int nCt = I >> 5;
  const int nMt = (O + 127) >> 7;
  const int OH = (stride == 2) ? H / 2 : H;
  const int OW = (stride == 2) ? W / 2 : W;
  const long iters = (long)B * OH * (OW >> 5);
  // target ~2048 blocks (2 WGs/CU + tail balance); atomics scale with
  // nsplit so cap at 128
  int nsplit = (int)((2048 + (long)nCt * nMt - 1) / ((long)nCt * nMt));
  if (nsplit > iters) nsplit = (int)iters;
  if (nsplit > 128) nsplit = 128;
  if (nsplit < 1) nsplit = 1;
  return nsplit;
}

void launch_conv2d_wgrad_slab_bf16(float* ws, const bf16* x, const bf16* dy,
                                   int B, int I, int H, int W, int O,
                                   int nsplit, int stride, hipStream_t s) {
  const int nCt = I >> 5;
  const int nMt = (O + 127) >> 7;
  // pair ids run 0..nMt*nsplit-1; grid.x covers xcd-slot * ct * pairHi
  const int npair = nMt * nsplit;
  const int pair_hi = (npair + 7) >> 3;
  dim3 grid((unsigned)(8 * nCt * pair_hi));
  if (stride == 1)
    hipLaunchKernelGGL(conv2d_wgrad_slab_bf16, grid, dim3(256), 0, s, ws, x,
                       dy, B, I, H, W, O, nsplit);
  else
    hipLaunchKernelGGL(conv2d_wgrad_slab_s2_bf16, grid, dim3(256), 0, s, ws,
                       x, dy, B, I, H, W, O, nsplit);
}

}  // namespace gfa
