// Weight repack kernels: one launch replacing the per-conv-call
// torch permute+contiguous (+zero-pad) chains that showed up as ~660
// small launches per training step in the r02 torch.profiler trace.
#include "common.h"

namespace gfa {

// w [O,I,3,3] -> blocked [mT][cB][9][128][32], O zero-padded to 128
// (the fwd slab layout; see conv2d_slab.hip header).
template <typename T>
__global__ void pack_w_blocked_kernel(T* __restrict__ out,
                                      const T* __restrict__ w, int O, int I,
                                      int mT) {
  const int cB = I >> 5;
  const long total = (long)mT * cB * 9 * 128 * 32;
  for (long idx = global_tid(); idx < total; idx += global_stride()) {
    int cc = (int)(idx & 31);
    long r = idx >> 5;
    int oo = (int)(r & 127);
    r >>= 7;
    int tap = (int)(r % 9);
    r /= 9;
    int cb = (int)(r % cB);
    int mt = (int)(r / cB);
    int o = mt * 128 + oo, i = cb * 32 + cc;
    out[idx] = (o < O) ? w[((long)o * I + i) * 9 + tap] : from_f32<T>(0.f);
  }
}

// w [O,I,kh,kw] -> [O][kh*kw][I] (the s2/up2 tap-major layout).
template <typename T>
__global__ void pack_w_o9i_kernel(T* __restrict__ out, const T* __restrict__ w,
                                  int O, int I, int KK) {
  const long total = (long)O * KK * I;
  for (long idx = global_tid(); idx < total; idx += global_stride()) {
    int i = (int)(idx % I);
    long r = idx / I;
    int tap = (int)(r % KK);
    int o = (int)(r / KK);
    out[idx] = w[((long)o * I + i) * KK + tap];
  }
}

template <typename T>
void launch_pack_w_blocked(T* out, const T* w, int O, int I, int mT,
                           hipStream_t s) {
  long total = (long)mT * (I >> 5) * 9 * 128 * 32;
  hipLaunchKernelGGL(pack_w_blocked_kernel<T>, stream_grid(total), dim3(256),
                     0, s, out, w, O, I, mT);
}
template <typename T>
void launch_pack_w_o9i(T* out, const T* w, int O, int I, int KK,
                       hipStream_t s) {
  long total = (long)O * KK * I;
  hipLaunchKernelGGL(pack_w_o9i_kernel<T>, stream_grid(total), dim3(256), 0,
                     s, out, w, O, I, KK);
}

template void launch_pack_w_blocked<bf16>(bf16*, const bf16*, int, int, int,
                                          hipStream_t);
template void launch_pack_w_blocked<float>(float*, const float*, int, int,
                                           int, hipStream_t);
template void launch_pack_w_o9i<bf16>(bf16*, const bf16*, int, int, int,
                                      hipStream_t);
template void launch_pack_w_o9i<float>(float*, const float*, int, int, int,
                                       hipStream_t);

}  // namespace gfa
