// Tap-major LDS-slab implicit-GEMM 3x3 conv for gfx950 (stride 1, pad 1).
//
// The generic kernel in conv2d.hip decodes 9 filter taps per K-step and
// re-reads every input pixel up to 9 times from global memory; its
// MFMA:staging ratio makes it staging-bound (~9% of bf16 MFMA peak,
// profiles/r01_summary.md). This kernel reorders K as (tap, channel):
// the 16x8-pixel output tile's input slab (18x10xBC, halo included) is
// staged in LDS ONCE per 32-channel block, and the 9 taps' B-operand
// fragments are read DIRECTLY from the slab at shifted offsets — no
// per-tap global traffic for B, no per-tap LDS re-staging.
//
//   C[o][p]  +=  sum_{t,c} W[o][t][c] * slab[(py+ty(t))(px+tx(t))][c]
//
// Weights are passed pre-packed in the BLOCKED layout
//   [mTile][cbTile][tap][128 o][32 c]   (o zero-padded to 128)
// so one wave's A fragment set for a (tap, cb) is a CONTIGUOUS 1 KB
// region: lanes 0..15 read rows o..o+15 at 16 B each, the four K
// granules the adjacent columns. The r01 [O][9][I] layout made each
// fragment a 16-way 16-B gather with stride 9*I — the strided L2
// traffic that bounded the kernel (A-constant experiment: 1169 vs 725
// TF/s, profiles/r01_summary.md). A-fragments still come from global
// (L2-hot), so the slab stays the only LDS producer and the tap loop
// runs with NO barriers; one __syncthreads per 32-channel block. The
// next channel block's slab is prefetched into the other buffer during
// the MFMA of the current one.
//
// Tiles (template TH): TH=16 -> 128(O) x 256(px as 16x16), each wave a
// 32x256 sub-tile (2x16 fragments; every A fragment reused 16x, no O row
// loaded twice — the strided weight-fragment traffic is this kernel's
// bound). TH=8 -> 128 x 128, wave 64x64 (4x4 fragments), for H==8
// feature maps. v_mfma_f32_16x16x32_bf16 throughout.
//
// Replaces the reference's cuDNN 3x3 convs (TF 1.14).
#include "common.h"
#include <type_traits>

namespace gfa {

namespace {
constexpr int SLAB_PIX = 40;           // u16 per slab pixel (32 + 8 pad)
}  // namespace

// TH = output-tile height (8 or 16). TH=16 gives each A fragment 8 B-
// fragments of reuse instead of 4, halving the strided A-fragment
// traffic that bounds this kernel (an A-constant experiment ran 2.25x
// faster), at the cost of acc[4][8] register pressure.
template <int TH>
__global__ __launch_bounds__(256, TH == 16 ? 1 : 2)
void conv2d_fwd_slab_bf16(bf16* __restrict__ y, const bf16* __restrict__ x,
                          const bf16* __restrict__ wr,  // blocked, see top
                          int B, int I, int H, int W, int O) {
  // TH=8: wave = 64(O) x 64(px), acc 4x4. TH=16: wave = 32(O) x 256(px),
  // acc 2x16 — each A fragment reused 16x and no row is loaded by two
  // waves, quartering the strided A traffic per output.
  constexpr int MI = (TH == 16) ? 2 : 4;       // mi frags per wave
  constexpr int NI = (TH == 16) ? 16 : TH / 2; // ni frags per wave
  constexpr int SROWS = TH + 2;
  constexpr int NTASK = SROWS * 32;
  __shared__ u16 slab[2][SROWS * 18 * SLAB_PIX];

  const int tilesX = W >> 4;
  const int b = blockIdx.z;
  const int m0 = blockIdx.y * 128;
  const int ty = blockIdx.x / tilesX, tx = blockIdx.x % tilesX;
  const int y0 = ty * TH, x0 = tx * 16;

  const bf16* xb = x + (long)b * I * H * W;
  bf16* yb = y + (long)b * O * H * W;

  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const int wm = (TH == 16) ? wave * 32 : (wave >> 1) * 64;
  const int wpy = (TH == 16) ? 0 : (wave & 1) * NI;
  const int px = lane & 15;               // B-fragment pixel x (fixed)
  const int ag = lane >> 4;               // K granule 0..3

  const int nCB = I >> 5;

  // A-fragment bases in the blocked layout: row inside this m-tile's
  // 128-padded block (out-of-range rows are ZERO-padded by the packer,
  // so no clamp/predicate needed) plus the K granule column. The
  // (cb, tap) block offset is added in the loop: each tap block is
  // 128*32 = 4096 elements. Offsets fit int32 (mT*nCB*9*4096 <= 2.4M
  // + padding).
  int a_off[MI];
#pragma unroll
  for (int mi = 0; mi < MI; ++mi) {
    const int orow = wm + mi * 16 + (lane & 15);
    a_off[mi] = blockIdx.y * nCB * 9 * 4096 + orow * 32 + ag * 8;
  }

  // Slab staging is split into a LOAD phase (global -> registers, issued
  // at the top of a channel block) and a WRITE phase (registers -> LDS,
  // after the tap loop): a combined load+write forces the s_waitcnt for
  // HBM data before the block's first MFMA and parks the wave (~85%
  // SQ_WAIT_ANY measured). Each thread owns <=2 (dy, c) row tasks of 24
  // elements held as 6 s16x4.
  constexpr int KMAX = (NTASK + 255) / 256;
  const int n_task = (t < NTASK - (KMAX - 1) * 256) ? KMAX : KMAX - 1;
  s16x4 tk[KMAX][6];

  auto slab_load = [&](int c0) {
#pragma unroll
    for (int k = 0; k < KMAX; ++k) {
      if (k >= n_task) break;
      const int idx = t + k * 256;
      const int dy = idx >> 5, c = idx & 31;
      const int gy = y0 + dy - 1;
      const bf16* src = xb + ((long)(c0 + c) * H + gy) * W;
      if (gy < 0 || gy >= H) {
#pragma unroll
        for (int j = 0; j < 6; ++j) tk[k][j] = s16x4{};
      } else if (x0 >= 4 && x0 + 20 <= W) {
        // interior: 6 aligned 8-B loads covering [x0-4, x0+20)
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
    for (int k = 0; k < KMAX; ++k) {
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

  f32x4 acc[MI][NI] = {};

  slab_load(0);
  slab_write(0);

  __syncthreads();

  // A-fragments load at use (L2-hot weights; 3 waves/SIMD hide each
  // other's latency). Slab staging for the next channel block: loads
  // issue at the top of the block, the LDS write lands after tap 1 so
  // the register live-range of the staged rows stays short while the
  // global latency still hides behind two taps of MFMA.
  // TH=16: A fragments are software-pipelined one tap ahead in
  // registers (8 VGPRs per buffer at MI=2 — affordable where the
  // narrow tile's MI=4 version spilled); the loads for tap t+1 issue
  // before tap t's MFMA block and cover the L2/L3 latency.
  s16x8 afp[MI];
  if (TH == 16) {
#pragma unroll
    for (int mi = 0; mi < MI; ++mi)
      afp[mi] = *reinterpret_cast<const s16x8*>(wr + a_off[mi]);
  }

  for (int cb = 0; cb < nCB; ++cb) {
    const int c0 = cb << 5;
    const int sb = cb & 1;
    const bool pre = cb + 1 < nCB;
    if (pre) slab_load(c0 + 32);
#pragma unroll
    for (int tap = 0; tap < 9; ++tap) {
      const int r = tap / 3, s = tap - r * 3;
      const int tc = (cb * 9 + tap) * 4096;
      s16x8 af[MI];
      if (TH == 16) {
#pragma unroll
        for (int mi = 0; mi < MI; ++mi) af[mi] = afp[mi];
        const bool more = tap < 8 || pre;
        if (more) {
          const int ntc = tap < 8 ? tc + 4096 : (cb + 1) * 9 * 4096;
#pragma unroll
          for (int mi = 0; mi < MI; ++mi)
            afp[mi] = *reinterpret_cast<const s16x8*>(wr + a_off[mi] + ntc);
        }
      } else {
#pragma unroll
        for (int mi = 0; mi < MI; ++mi)
          af[mi] = *reinterpret_cast<const s16x8*>(wr + a_off[mi] + tc);
      }
#pragma unroll
      for (int ni = 0; ni < NI; ++ni) {
        const int py = wpy + ni;
        const s16x8 bfr = *reinterpret_cast<const s16x8*>(
            &slab[sb][((py + r) * 18 + px + s) * SLAB_PIX + (ag << 3)]);
#pragma unroll
        for (int mi = 0; mi < MI; ++mi)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mi], bfr, acc[mi][ni], 0, 0, 0);
      }
      if (tap == 1 && pre) slab_write(sb ^ 1);
    }
    __syncthreads();
  }

  // ---- epilogue: scalar bf16 stores (stride H*W between o rows) ----
#pragma unroll
  for (int ni = 0; ni < NI; ++ni) {
    const int gy = y0 + wpy + ni;
    const int gx = x0 + px;
    bf16* yp = yb + (long)gy * W + gx;
#pragma unroll
    for (int mi = 0; mi < MI; ++mi) {
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
  if ((H & 15) == 0 && H >= 16) {
    dim3 grid((W >> 4) * (H >> 4), ceil_div(O, 128), B);
    hipLaunchKernelGGL(conv2d_fwd_slab_bf16<16>, grid, dim3(256), 0, s, y,
                       x, wr, B, I, H, W, O);
  } else {
    dim3 grid((W >> 4) * (H >> 3), ceil_div(O, 128), B);
    hipLaunchKernelGGL(conv2d_fwd_slab_bf16<8>, grid, dim3(256), 0, s, y, x,
                       wr, B, I, H, W, O);
  }
}

}  // namespace gfa
