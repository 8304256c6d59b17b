// Common helpers for the gansformer_amd gfx950 HIP kernels.
// Target: MI355X (CDNA4, wave64, MFMA). No CUDA compatibility paths.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define GFA_DEV __device__ __forceinline__

namespace gfa {

using bf16 = __hip_bfloat16;

typedef float  f32x4  __attribute__((ext_vector_type(4)));
typedef float  f32x16 __attribute__((ext_vector_type(16)));
typedef short  s16x4  __attribute__((ext_vector_type(4)));
typedef short  s16x8  __attribute__((ext_vector_type(8)));
typedef unsigned short u16;

// ---- scalar dtype conversion ----
GFA_DEV float to_f32(float v) { return v; }
GFA_DEV float to_f32(bf16 v) { return __bfloat162float(v); }
GFA_DEV float bf16_bits_to_f32(u16 bits) {
  union { unsigned int u; float f; } cvt;
  cvt.u = (unsigned int)bits << 16;
  return cvt.f;
}
GFA_DEV u16 f32_to_bf16_bits(float v) {
  // round-to-nearest-even
  union { float f; unsigned int u; } cvt;
  cvt.f = v;
  unsigned int lsb = (cvt.u >> 16) & 1u;
  unsigned int rounded = cvt.u + 0x7FFFu + lsb;
  return (u16)(rounded >> 16);
}

template <typename T> GFA_DEV T from_f32(float v);
template <> GFA_DEV float from_f32<float>(float v) { return v; }
template <> GFA_DEV bf16 from_f32<bf16>(float v) {
  return __float2bfloat16(v);
}

// raw (bit-level) storage type: __hip_bfloat16 is a struct and cannot be
// an ext_vector element, so vectorized kernels work on u16 bits.
template <typename T> struct Raw { using type = T; };
template <> struct Raw<bf16> { using type = u16; };

GFA_DEV float raw_to_f32(float v) { return v; }
GFA_DEV float raw_to_f32(u16 v) { return bf16_bits_to_f32(v); }
template <typename R> GFA_DEV R f32_to_raw(float v);
template <> GFA_DEV float f32_to_raw<float>(float v) { return v; }
template <> GFA_DEV u16 f32_to_raw<u16>(float v) {
  return f32_to_bf16_bits(v);
}

// ---- shared LDS tile machinery for MFMA kernels ----
// [row][BK] tiles; bf16 uses an 8-element XOR swizzle so 16-byte fragment
// reads are bank-spread (guide T2); f32 uses +1 padding. mfma() performs
// one full-BK accumulate for a 16x16 fragment pair.
template <typename T> struct TileOps;

template <> struct TileOps<bf16> {
  // Row stride 40 u16 = 80 B = 20 dwords: rows 0..15 start at 16 distinct
  // bank quads ((row*20) % 64 is a bijection onto {0,4,...,60}), so the
  // 16-lane ds_read_b128 fragment reads are conflict-free, and every row
  // base stays 16-B aligned (Guideline 17). Scalar b16 writes are <=2-way.
  static constexpr int BK = 32;   // K depth per tile
  static constexpr int ROW = 40;  // u16 elements per LDS row (32 + 8 pad)
  using elem = u16;
  GFA_DEV static int idx(int row, int k) { return row * ROW + k; }
  GFA_DEV static void store(elem* lds, int row, int k, float v) {
    lds[row * ROW + k] = f32_to_bf16_bits(v);
  }
  // store a full 8-element granule (g = granule index in [0, 4))
  GFA_DEV static void store_vec8(elem* lds, int row, int g, s16x8 v) {
    *reinterpret_cast<s16x8*>(&lds[row * ROW + (g << 3)]) = v;
  }
  GFA_DEV static f32x4 mfma(const elem* As, const elem* Bs, int arow0,
                            int brow0, int lane, f32x4 acc) {
    int ar = arow0 + (lane & 15);
    int br = brow0 + (lane & 15);
    int g = lane >> 4;
    s16x8 af = *reinterpret_cast<const s16x8*>(&As[ar * ROW + (g << 3)]);
    s16x8 bfr = *reinterpret_cast<const s16x8*>(&Bs[br * ROW + (g << 3)]);
    return __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bfr, acc, 0, 0, 0);
  }
};

template <> struct TileOps<float> {
  static constexpr int BK = 32;
  static constexpr int ROW = 33;  // +1 pad breaks bank conflicts
  using elem = float;
  GFA_DEV static int idx(int row, int k) { return row * ROW + k; }
  GFA_DEV static void store(elem* lds, int row, int k, float v) {
    lds[idx(row, k)] = v;
  }
  GFA_DEV static f32x4 mfma(const elem* As, const elem* Bs, int arow0,
                            int brow0, int lane, f32x4 acc) {
    int ar = arow0 + (lane & 15);
    int br = brow0 + (lane & 15);
#pragma unroll
    for (int kk = 0; kk < BK; kk += 4) {
      float av = As[ar * ROW + kk + (lane >> 4)];
      float bv = Bs[br * ROW + kk + (lane >> 4)];
      acc = __builtin_amdgcn_mfma_f32_16x16x4f32(av, bv, acc, 0, 0, 0);
    }
    return acc;
  }
};

// ---- shared MFMA-operand staging (attention kernels) ----
// stage a [64 rows x BK] tile from row-major global memory: rows >=
// vrows and columns >= cmax zero-filled. bf16 path moves 16-B s16x8
// granules as raw bits; f32 falls back to scalar stores. Caller must
// pass vec8 = (strides % 8 == 0) for the vector path. One call covers
// exactly the 256-thread block's work; __syncthreads is the caller's.
template <typename T>
GFA_DEV void stage_tile_rows(int t, typename TileOps<T>::elem* buf,
                             const T* src, long rstride, int vrows, int c0,
                             int cmax, bool vec8) {
  using TO = TileOps<T>;
  constexpr int BK = TO::BK;
  if constexpr (sizeof(typename TO::elem) == 2) {
    if (vec8) {
      const int row = t >> 2, g = t & 3;  // 256 tasks exactly
      const int c = c0 + g * 8;
      s16x8 val{};
      if (row < vrows && c < cmax)
        val = *reinterpret_cast<const s16x8*>(src + row * rstride + c);
      TO::store_vec8(buf, row, g, val);
      return;
    }
  }
  for (int i = t; i < 64 * BK; i += 256) {
    int row = i / BK, c = c0 + i % BK;
    float v_ = 0.f;
    if (row < vrows && c < cmax) v_ = to_f32(src[row * rstride + c]);
    TO::store(buf, row, i % BK, v_);
  }
}

// stage a transposed [64 global-cols x 64 src-rows] tile PAIR:
// buf rows = global column index (c0 + 0..63), depth = source row
// (two BK tiles back-to-back at stride 64*ROW).
template <typename T>
GFA_DEV void stage_tile_trans(int t, typename TileOps<T>::elem* buf,
                              const T* src, long rstride, int vrows, int c0,
                              int cmax, bool vec8) {
  using TO = TileOps<T>;
  constexpr int BK = TO::BK;
  constexpr int ROW = TO::ROW;
  if constexpr (sizeof(typename TO::elem) == 2) {
    if (vec8) {
      for (int task = t; task < 512; task += 256) {
        const int srow = task >> 3, g = task & 7;
        const int c = c0 + g * 8;
        s16x8 val{};
        if (srow < vrows && c < cmax)
          val = *reinterpret_cast<const s16x8*>(src + srow * rstride + c);
        const u16* pv = reinterpret_cast<const u16*>(&val);
        auto* tb = buf + (srow / BK) * 64 * ROW;
        const int kk = srow % BK;
#pragma unroll
        for (int j = 0; j < 8; ++j) tb[(g * 8 + j) * ROW + kk] = pv[j];
      }
      return;
    }
  }
  for (int i = t; i < 64 * 64; i += 256) {
    int srow = i >> 6, c = c0 + (i & 63);
    float v_ = 0.f;
    if (srow < vrows && c < cmax) v_ = to_f32(src[srow * rstride + c]);
    TO::store(buf + (srow / BK) * 64 * ROW, i & 63, srow % BK, v_);
  }
}

// ---- grid helpers ----
GFA_DEV long global_tid() {
  return (long)blockIdx.x * blockDim.x + threadIdx.x;
}
GFA_DEV long global_stride() {
  return (long)gridDim.x * blockDim.x;
}

inline int ceil_div(long a, long b) { return (int)((a + b - 1) / b); }

// memory-bound launch: cap grid, grid-stride the rest (guide G11)
inline dim3 stream_grid(long n, int block = 256, int max_blocks = 2048) {
  long blocks = (n + block - 1) / block;
  if (blocks > max_blocks) blocks = max_blocks;
  if (blocks < 1) blocks = 1;
  return dim3((unsigned)blocks);
}

}  // namespace gfa

#define HIP_CHECK_LAST()                                                    \
  do {                                                                      \
    hipError_t _e = hipGetLastError();                                      \
    if (_e != hipSuccess) {                                                 \
      TORCH_CHECK(false, "HIP kernel launch failed: ",                      \
                  hipGetErrorString(_e));                                   \
    }                                                                       \
  } while (0)
