// Tall-skinny MFMA GEMM for the attention/mapping projections
// (hipBLASLt picks ~70 TF/s
// algorithms for M ~ B*HW (10^5..10^6), N,K in 128..512).
//
// C[M,N] = A[M,K] . op(B), two B layouts:
//   transB = true : B row-major [N,K]  (y = x W^T, the FC forward)
//   transB = false: B row-major [K,N]  (dx = dy W, the FC dgrad)
//
// Shape regime: M huge -> grid parallelism comes from M; N,K <= ~512.
// For N=K=128 the kernel is A-traffic bound (intensity ~64 flop/B ->
// ~0.5 PF/s roofline at 8 TB/s HBM3E); block tile 128x128 reads A
// exactly once per N/128 column tiles. Double-buffered LDS staging
// (load chunk c+1 into registers while MFMAing chunk c from LDS).
#include "common.h"

namespace gfa {

struct GemmParams {
  long M;
  int N, K;
};

// grid: (ceil(M/128) * ceil(N/128)); block 256 (4 waves).
// wave w owns C rows [w*32, w*32+32) of the 128-row tile: 2 x 8 MFMA
// fragments (32 M x 128 N), 16 f32x4 accumulators.
template <typename T, bool TRANS_B>
__global__ __launch_bounds__(256)
void gemm_skinny_kernel(T* __restrict__ cm, const T* __restrict__ am,
                        const T* __restrict__ bm, GemmParams p) {
  using TO = TileOps<T>;
  constexpr int BK = TO::BK;
  constexpr int ROW = TO::ROW;
  using elem = typename TO::elem;
  // bf16: double-buffered (41 KB). f32: single buffer (double would
  // blow the 64 KB workgroup LDS limit; the f32 path only serves the
  // low-res fp32 blocks, where this kernel is not the bottleneck).
  constexpr int NBUF = (sizeof(elem) == 2) ? 2 : 1;
  __shared__ elem As[NBUF][128 * ROW];
  __shared__ elem Bs[NBUF][128 * ROW];

  const int ntiles = (p.N + 127) / 128;
  const long m0 = (long)(blockIdx.x / ntiles) * 128;
  const int n0 = (blockIdx.x % ntiles) * 128;
  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const int nchunks = (p.K + BK - 1) / BK;

  // ---- staging helpers (zero-pad beyond M/N/K) ----
  auto stage_a = [&](int buf, int k0) {
    for (int i = t; i < 128 * BK; i += 256) {
      int row = i / BK, kk = k0 + i % BK;
      float v = 0.f;
      if (m0 + row < p.M && kk < p.K)
        v = to_f32(am[(m0 + row) * p.K + kk]);
      TO::store(As[buf], row, i % BK, v);
    }
  };
  auto stage_b = [&](int buf, int k0) {
    if (TRANS_B) {
      // B [N,K]: tile rows = n, depth = k (coalesced over k)
      for (int i = t; i < 128 * BK; i += 256) {
        int row = i / BK, kk = k0 + i % BK;
        float v = 0.f;
        if (n0 + row < p.N && kk < p.K)
          v = to_f32(bm[(long)(n0 + row) * p.K + kk]);
        TO::store(Bs[buf], row, i % BK, v);
      }
    } else {
      // B [K,N]: tile rows = n, depth = k; read coalesced over n,
      // store transposed into LDS
      for (int i = t; i < 128 * BK; i += 256) {
        int kk = k0 + i / 128, col = i % 128;  // i/128 in [0,BK)
        float v = 0.f;
        if (kk < p.K && n0 + col < p.N)
          v = to_f32(bm[(long)kk * p.N + n0 + col]);
        TO::store(Bs[buf], col, i / 128, v);
      }
    }
  };

  f32x4 acc[2][8] = {};
  if (NBUF == 2) {
    stage_a(0, 0);
    stage_b(0, 0);
    __syncthreads();
    for (int c = 0; c < nchunks; ++c) {
      int cur = c & 1, nxt = cur ^ 1;
      if (c + 1 < nchunks) {
        stage_a(nxt % NBUF, (c + 1) * BK);
        stage_b(nxt % NBUF, (c + 1) * BK);
      }
#pragma unroll
      for (int mi = 0; mi < 2; ++mi)
#pragma unroll
        for (int ni = 0; ni < 8; ++ni)
          acc[mi][ni] = TO::mfma(As[cur % NBUF], Bs[cur % NBUF],
                                 wave * 32 + mi * 16, ni * 16, lane,
                                 acc[mi][ni]);
      __syncthreads();
    }
  } else {
    for (int c = 0; c < nchunks; ++c) {
      stage_a(0, c * BK);
      stage_b(0, c * BK);
      __syncthreads();
#pragma unroll
      for (int mi = 0; mi < 2; ++mi)
#pragma unroll
        for (int ni = 0; ni < 8; ++ni)
          acc[mi][ni] = TO::mfma(As[0], Bs[0], wave * 32 + mi * 16,
                                 ni * 16, lane, acc[mi][ni]);
      __syncthreads();
    }
  }
  // ---- write C ----
#pragma unroll
  for (int mi = 0; mi < 2; ++mi)
#pragma unroll
    for (int ni = 0; ni < 8; ++ni) {
      int n = n0 + ni * 16 + (lane & 15);
      if (n >= p.N) continue;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        long m = m0 + wave * 32 + mi * 16 + (lane >> 4) * 4 + reg;
        if (m < p.M) cm[m * p.N + n] = from_f32<T>(acc[mi][ni][reg]);
      }
    }
}

template <typename T>
void launch_gemm_skinny(T* c, const T* a, const T* b, const GemmParams& p,
                        bool trans_b, hipStream_t s) {
  int ntiles = (p.N + 127) / 128;
  long mtiles = (p.M + 127) / 128;
  dim3 grid((unsigned)(mtiles * ntiles));
  if (trans_b)
    hipLaunchKernelGGL((gemm_skinny_kernel<T, true>), grid, dim3(256), 0, s,
                       c, a, b, p);
  else
    hipLaunchKernelGGL((gemm_skinny_kernel<T, false>), grid, dim3(256), 0, s,
                       c, a, b, p);
}

template void launch_gemm_skinny<float>(float*, const float*, const float*,
                                        const GemmParams&, bool, hipStream_t);
template void launch_gemm_skinny<bf16>(bf16*, const bf16*, const bf16*,
                                       const GemmParams&, bool, hipStream_t);

}  // namespace gfa
