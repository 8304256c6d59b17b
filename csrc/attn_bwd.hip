// Fused bipartite-attention backward (SURVEY.md K5 bwd; replaces the
// eager recompute-over-hipBLASLt composition flagged in VERDICT r01 #4).
//
// Math (S = Q K^T * scale, A = softmax_k(S), O = A V), given dO:
//   dV = A^T dO
//   dA = dO V^T
//   dS = A .* (dA - rowsum(dA .* A)) * scale
//   dQ = dS K,   dK = dS^T Q
// rowsum(dA .* A) == rowsum(dO .* O) (the flash-attention identity), so
// the host passes drow = (dO*O).sum(-1) and the kernels never need a
// cross-tile reduction for it.
//
//  * small-N path (Simplex: Nq = HW large, Nk <= 64): one block per
//    64-query tile recomputes S -> A in LDS, then runs all five GEMMs
//    on MFMA from LDS tiles. dQ writes are tile-exclusive; dK/dV are
//    reductions over query tiles -> fp32 atomics into workspace.
//  * long-N path (Duplex reverse: Nq <= 64, Nk = HW large): one block
//    per 64-key chunk rebuilds its A_c slab from the forward's saved
//    (m, l) softmax stats, writes its own dK/dV rows directly, and
//    atomically accumulates dQ (small: [Nq, D]).
//
// All global staging is vectorized: bf16 rows move as 16-B s16x8
// granules (raw bit copies, no per-element float round-trips) — the
// first version's scalar 2-B loads made the kernel pure-issue-bound
// (~680 load instructions per thread). fp32 softmax/dS math in both
// paths; MFMA operands in the input dtype, accumulate fp32.
#include "common.h"

namespace gfa {

struct AttnBwdParams {
  int B, Nq, Nk, D, E;
  float scale;
};

// ---------------------------------------------------------------------------
// small-N backward: grid (ceil(Nq/64), B), block 256. Requires Nk <= 64.
// dq: [B,Nq,D] (dtype T, direct); dkw: [B,Nk,D] f32 (atomic);
// dvw: [B,Nk,E] f32 (atomic); drow: [B,Nq] f32.
// ---------------------------------------------------------------------------
template <typename T>
__global__ __launch_bounds__(256)
void attn_smalln_bwd(T* __restrict__ dq, float* __restrict__ dkw,
                     float* __restrict__ dvw, const T* __restrict__ q,
                     const T* __restrict__ k, const T* __restrict__ v,
                     const T* __restrict__ dout, const float* __restrict__ drow,
                     AttnBwdParams p) {
  using TO = TileOps<T>;
  constexpr int BK = TO::BK;
  constexpr int ROW = TO::ROW;
  using elem = typename TO::elem;
  constexpr bool BF = (sizeof(elem) == 2);
  __shared__ float Ss[64][65];          // A, later dS (stride 65 = 1 mod 64 banks: the 68 stride made every column-parallel access 4-way conflicted - 8.2 conflict cycles/LDS inst measured, gpurun_out/r02_attnpmc)
  __shared__ elem Ta[64 * 2 * ROW];     // staging pair (depth up to 64)
  __shared__ elem Tb[64 * 2 * ROW];

  const int b = blockIdx.y;
  const int q0 = blockIdx.x * 64;
  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const int nk_t16 = (p.Nk + 15) / 16;
  const int nk_bk = (p.Nk <= BK) ? 1 : 2;  // depth tiles when depth = keys
  // D/E are attention dims (multiples of 8 in every model config); the
  // vector path needs 16-B aligned rows
  const bool vec8 = BF && (p.D % 8 == 0) && (p.E % 8 == 0);

  const T* qb = q + ((long)b * p.Nq) * p.D;
  const T* kb = k + ((long)b * p.Nk) * p.D;
  const T* vb = v + ((long)b * p.Nk) * p.E;
  const T* dob = dout + ((long)b * p.Nq) * p.E;
  T* dqb = dq + ((long)b * p.Nq) * p.D;
  float* dkb = dkw + ((long)b * p.Nk) * p.D;
  float* dvb = dvw + ((long)b * p.Nk) * p.E;
  const float* drb = drow + (long)b * p.Nq;

  // staging delegates to the shared helpers in common.h
  auto stage_rows = [&](elem* buf, const T* src, long rstride, int vrows,
                        int c0, int cmax) {
    stage_tile_rows<T>(t, buf, src, rstride, vrows, c0, cmax, vec8);
  };
  auto stage_trans = [&](elem* buf, const T* src, long rstride, int vrows,
                         int c0, int cmax) {
    stage_tile_trans<T>(t, buf, src, rstride, vrows, c0, cmax, vec8);
  };

  // ---------- phase A: S = Q K^T -> softmax -> A in Ss ----------
  f32x4 acc_s[4] = {};
  for (int d0 = 0; d0 < p.D; d0 += BK) {
    stage_rows(Ta, qb + (long)q0 * p.D, p.D, p.Nq - q0, d0, p.D);
    stage_rows(Tb, kb, p.D, p.Nk, d0, p.D);
    __syncthreads();
#pragma unroll
    for (int ni = 0; ni < 4; ++ni)
      if (ni < nk_t16)
        acc_s[ni] = TO::mfma(Ta, Tb, wave * 16, ni * 16, lane, acc_s[ni]);
    __syncthreads();
  }
#pragma unroll
  for (int ni = 0; ni < 4; ++ni)
    if (ni < nk_t16)
#pragma unroll
      for (int reg = 0; reg < 4; ++reg)
        Ss[wave * 16 + (lane >> 4) * 4 + reg][ni * 16 + (lane & 15)] =
            acc_s[ni][reg];
  __syncthreads();
  if (t < 64) {
    float m = -3.0e38f;
    for (int j = 0; j < p.Nk; ++j) m = fmaxf(m, Ss[t][j] * p.scale);
    float l = 0.f;
    for (int j = 0; j < p.Nk; ++j) {
      float pv = __expf(Ss[t][j] * p.scale - m);
      l += pv;
      Ss[t][j] = pv;
    }
    float inv = 1.f / l;
    for (int j = 0; j < 64; ++j) Ss[t][j] = (j < p.Nk) ? Ss[t][j] * inv : 0.f;
  }
  __syncthreads();

  // ---------- phase B: dV += A^T dO (rows = keys, depth = q) ----------
  for (int i = t; i < 64 * 64; i += 256) {
    int key = i >> 6, qq = i & 63;
    TO::store(Ta + (qq / BK) * 64 * ROW, key, qq % BK, Ss[qq][key]);
  }
  for (int e0 = 0; e0 < p.E; e0 += 64) {
    stage_trans(Tb, dob + (long)q0 * p.E, p.E, p.Nq - q0, e0, p.E);
    __syncthreads();
    f32x4 accv[4] = {};
#pragma unroll
    for (int ei = 0; ei < 4; ++ei)
      for (int kk = 0; kk < 64; kk += BK)
        accv[ei] = TO::mfma(Ta + (kk / BK) * 64 * ROW,
                            Tb + (kk / BK) * 64 * ROW, wave * 16, ei * 16,
                            lane, accv[ei]);
#pragma unroll
    for (int ei = 0; ei < 4; ++ei) {
      int e = e0 + ei * 16 + (lane & 15);
      if (e >= p.E) continue;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        int key = wave * 16 + (lane >> 4) * 4 + reg;
        if (key < p.Nk) atomicAdd(&dvb[(long)key * p.E + e], accv[ei][reg]);
      }
    }
    __syncthreads();
  }

  // ---------- phase C: dA = dO V^T; dS = A.*(dA - drow)*scale ----------
  f32x4 acc_da[4] = {};
  for (int e0 = 0; e0 < p.E; e0 += BK) {
    stage_rows(Ta, dob + (long)q0 * p.E, p.E, p.Nq - q0, e0, p.E);
    stage_rows(Tb, vb, p.E, p.Nk, e0, p.E);
    __syncthreads();
#pragma unroll
    for (int ni = 0; ni < 4; ++ni)
      if (ni < nk_t16)
        acc_da[ni] = TO::mfma(Ta, Tb, wave * 16, ni * 16, lane, acc_da[ni]);
    __syncthreads();
  }
#pragma unroll
  for (int ni = 0; ni < 4; ++ni) {
    if (ni >= nk_t16) continue;
    int key = ni * 16 + (lane & 15);
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      int qq = wave * 16 + (lane >> 4) * 4 + reg;
      float dr = (q0 + qq < p.Nq) ? drb[q0 + qq] : 0.f;
      acc_da[ni][reg] = Ss[qq][key] * (acc_da[ni][reg] - dr) * p.scale;
    }
  }
  __syncthreads();  // all reads of A done before overwrite
#pragma unroll
  for (int ni = 0; ni < 4; ++ni)
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      int key = ni * 16 + (lane & 15);
      int qq = wave * 16 + (lane >> 4) * 4 + reg;
      Ss[qq][key] = (ni < nk_t16) ? acc_da[ni][reg] : 0.f;
    }
  __syncthreads();

  // ---------- phase D: dQ = dS K (rows = q, depth = key) ----------
  for (int i = t; i < 64 * 64; i += 256) {
    int qq = i >> 6, key = i & 63;
    TO::store(Ta + (key / BK) * 64 * ROW, qq, key % BK, Ss[qq][key]);
  }
  for (int d0 = 0; d0 < p.D; d0 += 64) {
    stage_trans(Tb, kb, p.D, p.Nk, d0, p.D);  // K^T: rows = d, depth = key
    __syncthreads();
    f32x4 accq[4] = {};
#pragma unroll
    for (int di = 0; di < 4; ++di)
      for (int kk = 0; kk < nk_bk * BK; kk += BK)
        accq[di] = TO::mfma(Ta + (kk / BK) * 64 * ROW,
                            Tb + (kk / BK) * 64 * ROW, wave * 16, di * 16,
                            lane, accq[di]);
#pragma unroll
    for (int di = 0; di < 4; ++di) {
      int dd = d0 + di * 16 + (lane & 15);
      if (dd >= p.D) continue;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        int qq = q0 + wave * 16 + (lane >> 4) * 4 + reg;
        if (qq < p.Nq) dqb[(long)qq * p.D + dd] = from_f32<T>(accq[di][reg]);
      }
    }
    __syncthreads();
  }

  // ---------- phase E: dK += dS^T Q (rows = key, depth = q) ----------
  for (int i = t; i < 64 * 64; i += 256) {
    int key = i >> 6, qq = i & 63;
    TO::store(Ta + (qq / BK) * 64 * ROW, key, qq % BK, Ss[qq][key]);
  }
  for (int d0 = 0; d0 < p.D; d0 += 64) {
    stage_trans(Tb, qb + (long)q0 * p.D, p.D, p.Nq - q0, d0, p.D);
    __syncthreads();
    f32x4 acck[4] = {};
#pragma unroll
    for (int di = 0; di < 4; ++di)
      for (int kk = 0; kk < 64; kk += BK)
        acck[di] = TO::mfma(Ta + (kk / BK) * 64 * ROW,
                            Tb + (kk / BK) * 64 * ROW, wave * 16, di * 16,
                            lane, acck[di]);
#pragma unroll
    for (int di = 0; di < 4; ++di) {
      int dd = d0 + di * 16 + (lane & 15);
      if (dd >= p.D) continue;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        int key = wave * 16 + (lane >> 4) * 4 + reg;
        if (key < p.Nk) atomicAdd(&dkb[(long)key * p.D + dd], acck[di][reg]);
      }
    }
    __syncthreads();
  }
}

// ---------------------------------------------------------------------------
// long-N backward: grid (nchunks, B), block 256. Requires Nq <= 64.
// dqw: [B,Nq,D] f32 (atomic); dk: [B,Nk,D] T (chunk-exclusive);
// dv: [B,Nk,E] T (chunk-exclusive); ml: [B,Nq,2] f32 from forward;
// drow: [B,Nq] f32.
// ---------------------------------------------------------------------------
template <typename T>
__global__ __launch_bounds__(256)
void attn_longn_bwd(float* __restrict__ dqw, T* __restrict__ dk,
                    T* __restrict__ dv, const T* __restrict__ q,
                    const T* __restrict__ k, const T* __restrict__ v,
                    const T* __restrict__ dout, const float* __restrict__ ml,
                    const float* __restrict__ drow, AttnBwdParams p) {
  using TO = TileOps<T>;
  constexpr int BK = TO::BK;
  constexpr int ROW = TO::ROW;
  using elem = typename TO::elem;
  constexpr bool BF = (sizeof(elem) == 2);
  __shared__ float Ss[64][65];          // A_c, later dS_c ([key][q]; stride 65, see small-N note)
  __shared__ elem Ta[64 * 2 * ROW];
  __shared__ elem Tb[64 * 2 * ROW];

  const int b = blockIdx.y;
  const int key0 = blockIdx.x * 64;
  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const int nq_t16 = (p.Nq + 15) / 16;
  const bool vec8 = BF && (p.D % 8 == 0) && (p.E % 8 == 0);

  const T* qb = q + ((long)b * p.Nq) * p.D;
  const T* kb = k + ((long)b * p.Nk) * p.D;
  const T* vb = v + ((long)b * p.Nk) * p.E;
  const T* dob = dout + ((long)b * p.Nq) * p.E;
  float* dqb = dqw + ((long)b * p.Nq) * p.D;
  T* dkb = dk + ((long)b * p.Nk) * p.D;
  T* dvb = dv + ((long)b * p.Nk) * p.E;

  auto stage_rows = [&](elem* buf, const T* src, long rstride, int vrows,
                        int c0, int cmax) {
    stage_tile_rows<T>(t, buf, src, rstride, vrows, c0, cmax, vec8);
  };
  auto stage_trans = [&](elem* buf, const T* src, long rstride, int vrows,
                         int c0, int cmax) {
    stage_tile_trans<T>(t, buf, src, rstride, vrows, c0, cmax, vec8);
  };

  // ---------- phase A: S_c = K_c Q^T -> A_c via saved (m, l) ----------
  f32x4 acc_s[4] = {};
  for (int d0 = 0; d0 < p.D; d0 += BK) {
    stage_rows(Ta, kb + (long)key0 * p.D, p.D, p.Nk - key0, d0, p.D);
    stage_rows(Tb, qb, p.D, p.Nq, d0, p.D);
    __syncthreads();
#pragma unroll
    for (int ni = 0; ni < 4; ++ni)
      if (ni < nq_t16)
        acc_s[ni] = TO::mfma(Ta, Tb, wave * 16, ni * 16, lane, acc_s[ni]);
    __syncthreads();
  }
#pragma unroll
  for (int ni = 0; ni < 4; ++ni)
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      int qq = ni * 16 + (lane & 15);
      int key = wave * 16 + (lane >> 4) * 4 + reg;
      Ss[key][qq] = (ni < nq_t16) ? acc_s[ni][reg] : 0.f;
    }
  __syncthreads();
  for (int i = t; i < 64 * 64; i += 256) {
    int key = i >> 6, qq = i & 63;  // each thread owns its element: no race
    float a = 0.f;
    if (key0 + key < p.Nk && qq < p.Nq) {
      float m_ = ml[((long)b * p.Nq + qq) * 2 + 0];
      float l_ = ml[((long)b * p.Nq + qq) * 2 + 1];
      a = __expf(Ss[key][qq] * p.scale - m_) / l_;
    }
    Ss[key][qq] = a;
  }
  __syncthreads();

  // ---------- phase B: dV_c = A_c dO (rows = key, depth = q) ----------
  for (int i = t; i < 64 * 64; i += 256) {
    int key = i >> 6, qq = i & 63;
    TO::store(Ta + (qq / BK) * 64 * ROW, key, qq % BK, Ss[key][qq]);
  }
  for (int e0 = 0; e0 < p.E; e0 += 64) {
    stage_trans(Tb, dob, p.E, p.Nq, e0, p.E);
    __syncthreads();
    f32x4 accv[4] = {};
#pragma unroll
    for (int ei = 0; ei < 4; ++ei)
      for (int kk = 0; kk < 64; kk += BK)
        accv[ei] = TO::mfma(Ta + (kk / BK) * 64 * ROW,
                            Tb + (kk / BK) * 64 * ROW, wave * 16, ei * 16,
                            lane, accv[ei]);
#pragma unroll
    for (int ei = 0; ei < 4; ++ei) {
      int e = e0 + ei * 16 + (lane & 15);
      if (e >= p.E) continue;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        int key = key0 + wave * 16 + (lane >> 4) * 4 + reg;
        if (key < p.Nk) dvb[(long)key * p.E + e] = from_f32<T>(accv[ei][reg]);
      }
    }
    __syncthreads();
  }

  // ---------- phase C: dA_c = V_c dO^T; dS_c = A.*(dA-drow)*scale ----------
  f32x4 acc_da[4] = {};
  for (int e0 = 0; e0 < p.E; e0 += BK) {
    stage_rows(Ta, vb + (long)key0 * p.E, p.E, p.Nk - key0, e0, p.E);
    stage_rows(Tb, dob, p.E, p.Nq, e0, p.E);
    __syncthreads();
#pragma unroll
    for (int ni = 0; ni < 4; ++ni)
      if (ni < nq_t16)
        acc_da[ni] = TO::mfma(Ta, Tb, wave * 16, ni * 16, lane, acc_da[ni]);
    __syncthreads();
  }
#pragma unroll
  for (int ni = 0; ni < 4; ++ni) {
    if (ni >= nq_t16) continue;
    int qq = ni * 16 + (lane & 15);
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      int key = wave * 16 + (lane >> 4) * 4 + reg;
      float dr = (qq < p.Nq) ? drow[(long)b * p.Nq + qq] : 0.f;
      acc_da[ni][reg] = Ss[key][qq] * (acc_da[ni][reg] - dr) * p.scale;
    }
  }
  __syncthreads();
#pragma unroll
  for (int ni = 0; ni < 4; ++ni)
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      int qq = ni * 16 + (lane & 15);
      int key = wave * 16 + (lane >> 4) * 4 + reg;
      Ss[key][qq] = (ni < nq_t16) ? acc_da[ni][reg] : 0.f;
    }
  __syncthreads();

  // ---------- phase D: dK_c = dS_c Q (rows = key, depth = q) ----------
  for (int i = t; i < 64 * 64; i += 256) {
    int key = i >> 6, qq = i & 63;
    TO::store(Ta + (qq / BK) * 64 * ROW, key, qq % BK, Ss[key][qq]);
  }
  for (int d0 = 0; d0 < p.D; d0 += 64) {
    stage_trans(Tb, qb, p.D, p.Nq, d0, p.D);  // Q^T: rows = d, depth = q
    __syncthreads();
    f32x4 acck[4] = {};
#pragma unroll
    for (int di = 0; di < 4; ++di)
      for (int kk = 0; kk < 64; kk += BK)
        acck[di] = TO::mfma(Ta + (kk / BK) * 64 * ROW,
                            Tb + (kk / BK) * 64 * ROW, wave * 16, di * 16,
                            lane, acck[di]);
#pragma unroll
    for (int di = 0; di < 4; ++di) {
      int dd = d0 + di * 16 + (lane & 15);
      if (dd >= p.D) continue;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        int key = key0 + wave * 16 + (lane >> 4) * 4 + reg;
        if (key < p.Nk) dkb[(long)key * p.D + dd] = from_f32<T>(acck[di][reg]);
      }
    }
    __syncthreads();
  }

  // ---------- phase E: dQ += dS_c^T K_c (rows = q, depth = key) ----------
  for (int i = t; i < 64 * 64; i += 256) {
    int qq = i >> 6, key = i & 63;
    TO::store(Ta + (key / BK) * 64 * ROW, qq, key % BK, Ss[key][qq]);
  }
  for (int d0 = 0; d0 < p.D; d0 += 64) {
    stage_trans(Tb, kb + (long)key0 * p.D, p.D, p.Nk - key0, d0, p.D);
    __syncthreads();
    f32x4 accq[4] = {};
#pragma unroll
    for (int di = 0; di < 4; ++di)
      for (int kk = 0; kk < 64; kk += BK)
        accq[di] = TO::mfma(Ta + (kk / BK) * 64 * ROW,
                            Tb + (kk / BK) * 64 * ROW, wave * 16, di * 16,
                            lane, accq[di]);
#pragma unroll
    for (int di = 0; di < 4; ++di) {
      int dd = d0 + di * 16 + (lane & 15);
      if (dd >= p.D) continue;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        int qq = wave * 16 + (lane >> 4) * 4 + reg;
        if (qq < p.Nq) atomicAdd(&dqb[(long)qq * p.D + dd], accq[di][reg]);
      }
    }
    __syncthreads();
  }
}

// ---------------- launchers ----------------
template <typename T>
void launch_attn_smalln_bwd(T* dq, float* dkw, float* dvw, const T* q,
                            const T* k, const T* v, const T* dout,
                            const float* drow, const AttnBwdParams& p,
                            hipStream_t s) {
  dim3 grid(ceil_div(p.Nq, 64), p.B);
  hipLaunchKernelGGL(attn_smalln_bwd<T>, grid, dim3(256), 0, s, dq, dkw, dvw,
                     q, k, v, dout, drow, p);
}
template <typename T>
void launch_attn_longn_bwd(float* dqw, T* dk, T* dv, const T* q, const T* k,
                           const T* v, const T* dout, const float* ml,
                           const float* drow, const AttnBwdParams& p,
                           hipStream_t s) {
  dim3 grid(ceil_div(p.Nk, 64), p.B);
  hipLaunchKernelGGL(attn_longn_bwd<T>, grid, dim3(256), 0, s, dqw, dk, dv, q,
                     k, v, dout, ml, drow, p);
}

template void launch_attn_smalln_bwd<float>(float*, float*, float*,
                                            const float*, const float*,
                                            const float*, const float*,
                                            const float*, const AttnBwdParams&,
                                            hipStream_t);
template void launch_attn_smalln_bwd<bf16>(bf16*, float*, float*, const bf16*,
                                           const bf16*, const bf16*,
                                           const bf16*, const float*,
                                           const AttnBwdParams&, hipStream_t);
template void launch_attn_longn_bwd<float>(float*, float*, float*,
                                           const float*, const float*,
                                           const float*, const float*,
                                           const float*, const float*,
                                           const AttnBwdParams&, hipStream_t);
template void launch_attn_longn_bwd<bf16>(float*, bf16*, bf16*, const bf16*,
                                          const bf16*, const bf16*,
                                          const bf16*, const float*,
                                          const float*, const AttnBwdParams&,
                                          hipStream_t);

}  // namespace gfa
