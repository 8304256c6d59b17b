// Bipartite attention: out = softmax(Q K^T * scale) V   (SURVEY.md K5).
// The GANsformer's signature op (TF matmul+softmax composition in the
// reference, ref src/training/networks.py [R]); here fused CDNA4 kernels:
//
//  * small-N path (N_kv <= 64; Simplex direction: HW image tokens attend
//    to k <= 32 latents): one kernel; K/V staged in LDS, Q tiled 64 rows
//    per block, QK^T and PV on MFMA, softmax over N_kv per query row.
//  * long-N path (Duplex reverse: k latents attend to HW tokens): 64-key
//    chunks compute partial (rowmax, sumexp, P.V) per block; a reduce
//    kernel combines chunks with max-rescaling (two-pass online softmax).
//
// bf16 path uses v_mfma_f32_16x16x32_bf16; f32 path (low-res fp32 blocks)
// uses the exact-f32 v_mfma_f32_16x16x4_f32. fp32 softmax in both.
#include "common.h"

namespace gfa {

// per-dtype LDS tile ops (TileOps) are shared in common.h.

struct AttnParams {
  int B, Nq, Nk, D, E;
  float scale;
};

// ---------------- small-N fused kernel ----------------
// grid: (ceil(Nq/64), B); block 256. Requires Nk <= 64.
template <typename T>
__global__ __launch_bounds__(256)
void attn_smalln_kernel(T* __restrict__ out, const T* __restrict__ q,
                        const T* __restrict__ k, const T* __restrict__ v,
                        AttnParams p) {
  using TO = TileOps<T>;
  constexpr int BK = TO::BK;
  constexpr int ROW = TO::ROW;
  using elem = typename TO::elem;
  __shared__ elem Qs[64 * ROW];
  __shared__ elem Ks[64 * ROW];
  __shared__ elem Ps[64 * (2 * ROW)];   // 64 q rows x 64 key cols (2 BK tiles)
  __shared__ elem Vt[64 * (2 * ROW)];   // 64 e rows x 64 key cols
  __shared__ float Ss[64][65];          // raw scores (stride 65 = 1 mod 64 banks: column-parallel access conflict-free)

  const int b = blockIdx.y;
  const int q0 = blockIdx.x * 64;
  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;              // wave w owns q rows [w*16, w*16+16)
  const int nk_t16 = (p.Nk + 15) / 16;  // fragment count over keys

  const T* qb = q + ((long)b * p.Nq) * p.D;
  const T* kb = k + ((long)b * p.Nk) * p.D;
  const T* vb = v + ((long)b * p.Nk) * p.E;
  T* ob = out + ((long)b * p.Nq) * p.E;

  const bool vecD = (sizeof(elem) == 2) && (p.D % 8 == 0);
  const bool vecE = (sizeof(elem) == 2) && (p.E % 8 == 0);
  // ---------- S = Q K^T ----------
  f32x4 acc_s[4] = {};
  for (int d0 = 0; d0 < p.D; d0 += BK) {
    stage_tile_rows<T>(t, Qs, qb + (long)q0 * p.D, p.D, p.Nq - q0, d0, p.D,
                       vecD);
    stage_tile_rows<T>(t, Ks, kb, p.D, p.Nk, d0, p.D, vecD);
    __syncthreads();
#pragma unroll
    for (int ni = 0; ni < 4; ++ni)
      if (ni < nk_t16)
        acc_s[ni] = TO::mfma(Qs, Ks, wave * 16, ni * 16, lane, acc_s[ni]);
    __syncthreads();
  }
  // write scores: C map col=lane&15, row=(lane>>4)*4+reg
#pragma unroll
  for (int ni = 0; ni < 4; ++ni)
    if (ni < nk_t16)
#pragma unroll
      for (int reg = 0; reg < 4; ++reg)
        Ss[wave * 16 + (lane >> 4) * 4 + reg][ni * 16 + (lane & 15)] =
            acc_s[ni][reg];
  __syncthreads();
  // ---------- softmax over Nk (one thread per query row) ----------
  if (t < 64) {
    float m = -3.0e38f;
    for (int j = 0; j < p.Nk; ++j)
      m = fmaxf(m, Ss[t][j] * p.scale);
    float l = 0.f;
    for (int j = 0; j < 64; ++j) {
      float pv = 0.f;
      if (j < p.Nk) {
        pv = __expf(Ss[t][j] * p.scale - m);
        l += pv;
      }
      // defer 1/l into the PV epilogue? fold it here: store normalized later
      Ss[t][j] = pv;
    }
    float inv = 1.f / l;
    for (int j = 0; j < 64; ++j) {
      float pv = Ss[t][j] * inv;
      TO::store(Ps + (j / BK) * 64 * ROW, t, j % BK, pv);
    }
  }
  __syncthreads();
  // ---------- O = P V ----------
  const int nk_pad = (p.Nk + BK - 1) / BK * BK;  // BK-multiple key depth
  for (int e0 = 0; e0 < p.E; e0 += 64) {
    // stage V^T tile pair: rows = e (64), depth = keys
    stage_tile_trans<T>(t, Vt, vb, p.E, p.Nk, e0, p.E, vecE);
    __syncthreads();
    f32x4 acc_o[4] = {};
#pragma unroll
    for (int ei = 0; ei < 4; ++ei) {
      for (int kk = 0; kk < nk_pad; kk += BK)
        acc_o[ei] = TO::mfma(Ps + (kk / BK) * 64 * ROW,
                             Vt + (kk / BK) * 64 * ROW,
                             wave * 16, ei * 16, lane, acc_o[ei]);
    }
    // write out
#pragma unroll
    for (int ei = 0; ei < 4; ++ei) {
      int e = e0 + ei * 16 + (lane & 15);
      if (e >= p.E) continue;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        int qq = q0 + wave * 16 + (lane >> 4) * 4 + reg;
        if (qq < p.Nq) ob[(long)qq * p.E + e] = from_f32<T>(acc_o[ei][reg]);
      }
    }
    __syncthreads();
  }
}

// ---------------- long-N partial kernel ----------------
// grid: (nchunks, B); chunk = 64 keys. Requires Nq <= 64.
// outputs: ws_m [B][nchunks][64], ws_l same, ws_o [B][nchunks][64][E] f32
template <typename T>
__global__ __launch_bounds__(256)
void attn_longn_partial(float* __restrict__ ws_m, float* __restrict__ ws_l,
                        float* __restrict__ ws_o, const T* __restrict__ q,
                        const T* __restrict__ k, const T* __restrict__ v,
                        AttnParams p) {
  using TO = TileOps<T>;
  constexpr int BK = TO::BK;
  constexpr int ROW = TO::ROW;
  using elem = typename TO::elem;
  __shared__ elem Ks[64 * ROW];
  __shared__ elem Qs[64 * ROW];
  __shared__ elem Pt[64 * (2 * ROW)];   // [q][64 keys]
  __shared__ elem Vt[64 * (2 * ROW)];   // [e][64 keys]
  __shared__ float SsT[64][65];         // [q][key] transposed scores (65: see Ss)

  const int b = blockIdx.y;
  const int c = blockIdx.x;             // chunk
  const int key0 = c * 64;
  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;              // wave owns key rows [w*16, ..)
  const int nchunks = gridDim.x;
  const int nq_t16 = (p.Nq + 15) / 16;

  const T* qb = q + ((long)b * p.Nq) * p.D;
  const T* kb = k + ((long)b * p.Nk) * p.D;
  const T* vb = v + ((long)b * p.Nk) * p.E;

  const bool vecD = (sizeof(elem) == 2) && (p.D % 8 == 0);
  const bool vecE = (sizeof(elem) == 2) && (p.E % 8 == 0);
  // ---------- S_c = K_c Q^T  (rows = keys, cols = queries) ----------
  f32x4 acc_s[4] = {};
  for (int d0 = 0; d0 < p.D; d0 += BK) {
    stage_tile_rows<T>(t, Ks, kb + (long)key0 * p.D, p.D, p.Nk - key0, d0,
                       p.D, vecD);
    stage_tile_rows<T>(t, Qs, qb, p.D, p.Nq, d0, p.D, vecD);
    __syncthreads();
#pragma unroll
    for (int ni = 0; ni < 4; ++ni)
      if (ni < nq_t16)
        acc_s[ni] = TO::mfma(Ks, Qs, wave * 16, ni * 16, lane, acc_s[ni]);
    __syncthreads();
  }
  // write transposed: SsT[q][key]
#pragma unroll
  for (int ni = 0; ni < 4; ++ni)
    if (ni < nq_t16)
#pragma unroll
      for (int reg = 0; reg < 4; ++reg)
        SsT[ni * 16 + (lane & 15)][wave * 16 + (lane >> 4) * 4 + reg] =
            acc_s[ni][reg];
  __syncthreads();
  // ---------- per-query chunk softmax partials ----------
  if (t < p.Nq) {
    float m = -3.0e38f;
    int valid = min(64, p.Nk - key0);
    for (int j = 0; j < valid; ++j)
      m = fmaxf(m, SsT[t][j] * p.scale);
    float l = 0.f;
    for (int j = 0; j < 64; ++j) {
      float pv = 0.f;
      if (j < valid) {
        pv = __expf(SsT[t][j] * p.scale - m);
        l += pv;
      }
      TO::store(Pt + (j / BK) * 64 * ROW, t, j % BK, pv);
    }
    ws_m[((long)b * nchunks + c) * 64 + t] = m;
    ws_l[((long)b * nchunks + c) * 64 + t] = l;
  } else if (t < 64) {
    for (int j = 0; j < 64; ++j)
      TO::store(Pt + (j / BK) * 64 * ROW, t, j % BK, 0.f);
  }
  __syncthreads();
  // ---------- O_c = P_c V_c ----------
  float* ob = ws_o + (((long)b * nchunks + c) * 64) * p.E;
  for (int e0 = 0; e0 < p.E; e0 += 64) {
    stage_tile_trans<T>(t, Vt, vb + (long)key0 * p.E, p.E, p.Nk - key0, e0,
                        p.E, vecE);
    __syncthreads();
    f32x4 acc_o[4] = {};
#pragma unroll
    for (int ei = 0; ei < 4; ++ei)
      for (int kk = 0; kk < 64; kk += BK)
        acc_o[ei] = TO::mfma(Pt + (kk / BK) * 64 * ROW,
                             Vt + (kk / BK) * 64 * ROW,
                             wave * 16, ei * 16, lane, acc_o[ei]);
#pragma unroll
    for (int ei = 0; ei < 4; ++ei) {
      int e = e0 + ei * 16 + (lane & 15);
      if (e >= p.E) continue;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        int qq = wave * 16 + (lane >> 4) * 4 + reg;
        if (qq < p.Nq) ob[(long)qq * p.E + e] = acc_o[ei][reg];
      }
    }
    __syncthreads();
  }
}

// ---------------- long-N reduce kernel ----------------
// grid: (Nq, B); block 256. Combines chunk partials with max-rescaling.
template <typename T>
__global__ void attn_longn_reduce(T* __restrict__ out,
                                  const float* __restrict__ ws_m,
                                  const float* __restrict__ ws_l,
                                  const float* __restrict__ ws_o,
                                  float* __restrict__ ml,  // optional [B,Nq,2]
                                  AttnParams p, int nchunks) {
  const int qq = blockIdx.x;
  const int b = blockIdx.y;
  const int t = threadIdx.x;
  __shared__ float red[256];
  __shared__ float m_glob, l_glob;
  // global max over chunks
  float m = -3.0e38f;
  for (int c = t; c < nchunks; c += 256)
    m = fmaxf(m, ws_m[((long)b * nchunks + c) * 64 + qq]);
  red[t] = m;
  __syncthreads();
  for (int s = 128; s > 0; s >>= 1) {
    if (t < s) red[t] = fmaxf(red[t], red[t + s]);
    __syncthreads();
  }
  if (t == 0) m_glob = red[0];
  __syncthreads();
  m = m_glob;
  float l = 0.f;
  for (int c = t; c < nchunks; c += 256) {
    long i = ((long)b * nchunks + c) * 64 + qq;
    l += ws_l[i] * __expf(ws_m[i] - m);
  }
  red[t] = l;
  __syncthreads();
  for (int s = 128; s > 0; s >>= 1) {
    if (t < s) red[t] += red[t + s];
    __syncthreads();
  }
  if (t == 0) l_glob = red[0];
  __syncthreads();
  if (t == 0 && ml != nullptr) {
    // saved for the fused backward: A = exp(S*scale - m) / l
    ml[((long)b * p.Nq + qq) * 2 + 0] = m_glob;
    ml[((long)b * p.Nq + qq) * 2 + 1] = l_glob;
  }
  float inv = 1.f / l_glob;
  for (int e = t; e < p.E; e += 256) {
    float acc = 0.f;
    for (int c = 0; c < nchunks; ++c) {
      long base = ((long)b * nchunks + c) * 64 + qq;
      acc += ws_o[base * p.E + e] * __expf(ws_m[base] - m);
    }
    out[((long)b * p.Nq + qq) * p.E + e] = from_f32<T>(acc * inv);
  }
}

// ---------------- launchers ----------------
template <typename T>
void launch_attn_smalln(T* out, const T* q, const T* k, const T* v,
                        const AttnParams& p, hipStream_t s) {
  dim3 grid(ceil_div(p.Nq, 64), p.B);
  hipLaunchKernelGGL(attn_smalln_kernel<T>, grid, dim3(256), 0, s, out, q, k,
                     v, p);
}
template <typename T>
void launch_attn_longn(T* out, const T* q, const T* k, const T* v,
                       float* ws_m, float* ws_l, float* ws_o, float* ml,
                       const AttnParams& p, int nchunks, hipStream_t s) {
  dim3 grid1(nchunks, p.B);
  hipLaunchKernelGGL(attn_longn_partial<T>, grid1, dim3(256), 0, s, ws_m,
                     ws_l, ws_o, q, k, v, p);
  dim3 grid2(p.Nq, p.B);
  hipLaunchKernelGGL(attn_longn_reduce<T>, grid2, dim3(256), 0, s, out, ws_m,
                     ws_l, ws_o, ml, p, nchunks);
}

template void launch_attn_smalln<float>(float*, const float*, const float*,
                                        const float*, const AttnParams&,
                                        hipStream_t);
template void launch_attn_smalln<bf16>(bf16*, const bf16*, const bf16*,
                                       const bf16*, const AttnParams&,
                                       hipStream_t);
template void launch_attn_longn<float>(float*, const float*, const float*,
                                       const float*, float*, float*, float*,
                                       float*, const AttnParams&, int,
                                       hipStream_t);
template void launch_attn_longn<bf16>(bf16*, const bf16*, const bf16*,
                                      const bf16*, float*, float*, float*,
                                      float*, const AttnParams&, int,
                                      hipStream_t);

}  // namespace gfa
