// Minibatch stddev statistics (discriminator; TF graph ops
// in the reference). Computes, per (batch-chunk m, feature-group f),
//   mean over (c, h, w) of sqrt(var over group members + eps)
// where the group of chunk m is {b = g*(B/G) + m : g in [0, G)}.
// One workgroup per (m, f); two-pass Welford-free reduction in LDS.
#include "common.h"

namespace gfa {

template <typename T>
__global__ void mbstd_kernel(float* __restrict__ out, const T* __restrict__ x,
                             int B, int C, int HW, int G, int F, float eps) {
  const int M = B / G;        // chunks
  const int c_per_f = C / F;
  const int m = blockIdx.x;   // chunk
  const int f = blockIdx.y;   // feature group
  const long n_items = (long)c_per_f * HW;

  __shared__ float red[256];
  float acc = 0.f;
  for (long it = threadIdx.x; it < n_items; it += blockDim.x) {
    int c = f * c_per_f + (int)(it / HW);
    int hw = (int)(it % HW);
    // mean over the G group members
    float mean = 0.f;
    float vals[32];  // G <= 32
    for (int g = 0; g < G; ++g) {
      int b = g * M + m;
      float v = to_f32(x[((long)b * C + c) * HW + hw]);
      vals[g] = v;
      mean += v;
    }
    mean /= G;
    float var = 0.f;
    for (int g = 0; g < G; ++g) {
      float d = vals[g] - mean;
      var += d * d;
    }
    var /= G;
    acc += sqrtf(var + eps);
  }
  red[threadIdx.x] = acc;
  __syncthreads();
  for (int s = blockDim.x / 2; s > 0; s >>= 1) {
    if (threadIdx.x < s) red[threadIdx.x] += red[threadIdx.x + s];
    __syncthreads();
  }
  if (threadIdx.x == 0) out[m * F + f] = red[0] / (float)n_items;
}

template <typename T>
void launch_mbstd(float* out, const T* x, int B, int C, int HW, int G, int F,
                  float eps, hipStream_t s) {
  dim3 grid(B / G, F);
  hipLaunchKernelGGL(mbstd_kernel<T>, grid, dim3(256), 0, s, out, x, B, C,
                     HW, G, F, eps);
}

template void launch_mbstd<float>(float*, const float*, int, int, int, int,
                                  int, float, hipStream_t);
template void launch_mbstd<bf16>(float*, const bf16*, int, int, int, int,
                                 int, float, hipStream_t);

}  // namespace gfa
