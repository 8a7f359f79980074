// Fused LayerNorm forward for gfx950: one workgroup per row, vectorized
// 16 B/lane loads, wave shuffle + LDS cross-wave reduction, single pass
// (sum + sumsq in f32), fused affine. Memory-bound — target is HBM BW.
#include "vfa_common.h"

namespace {

template <typename T>
__global__ void layernorm_kernel(const T* __restrict__ in,
                                 const T* __restrict__ weight,
                                 const T* __restrict__ bias,
                                 T* __restrict__ out, int rows, int d,
                                 float eps) {
  constexpr int VEC = 16 / sizeof(T);
  __shared__ float red[2][8];   // per-wave partials (<=8 waves of 64)
  const int row = blockIdx.x;
  if (row >= rows) return;
  const T* x = in + (long long)row * d;
  T* y = out + (long long)row * d;
  const int tid = threadIdx.x;
  const int lane = tid & 63, wave = tid >> 6;
  const int nwaves = blockDim.x >> 6;

  float sum = 0.f, sumsq = 0.f;
  using VecT = __attribute__((ext_vector_type(4))) unsigned;
  const int dvec = d / VEC;
  for (int i = tid; i < dvec; i += blockDim.x) {
    T tmp[VEC];
    *reinterpret_cast<VecT*>(tmp) =
        *reinterpret_cast<const VecT*>(x + (long long)i * VEC);
#pragma unroll
    for (int j = 0; j < VEC; ++j) {
      float v = to_f32<T>(tmp[j]);
      sum += v;
      sumsq += v * v;
    }
  }
  for (int i = dvec * VEC + tid; i < d; i += blockDim.x) {
    float v = to_f32<T>(x[i]);
    sum += v;
    sumsq += v * v;
  }
  sum = wave_reduce_sum(sum);
  sumsq = wave_reduce_sum(sumsq);
  if (lane == 0) { red[0][wave] = sum; red[1][wave] = sumsq; }
  __syncthreads();
  if (tid == 0) {
    float s = 0.f, ss = 0.f;
    for (int w = 0; w < nwaves; ++w) { s += red[0][w]; ss += red[1][w]; }
    float mean = s / d;
    float var = ss / d - mean * mean;
    red[0][0] = mean;
    red[1][0] = rsqrtf(var + eps);
  }
  __syncthreads();
  const float mean = red[0][0], rstd = red[1][0];

  for (int i = tid; i < dvec; i += blockDim.x) {
    T tx[VEC], tw[VEC], tb[VEC];
    *reinterpret_cast<VecT*>(tx) =
        *reinterpret_cast<const VecT*>(x + (long long)i * VEC);
    *reinterpret_cast<VecT*>(tw) =
        *reinterpret_cast<const VecT*>(weight + (long long)i * VEC);
    *reinterpret_cast<VecT*>(tb) =
        *reinterpret_cast<const VecT*>(bias + (long long)i * VEC);
#pragma unroll
    for (int j = 0; j < VEC; ++j) {
      float v = (to_f32<T>(tx[j]) - mean) * rstd;
      tx[j] = from_f32<T>(v * to_f32<T>(tw[j]) + to_f32<T>(tb[j]));
    }
    *reinterpret_cast<VecT*>(y + (long long)i * VEC) =
        *reinterpret_cast<VecT*>(tx);
  }
  for (int i = dvec * VEC + tid; i < d; i += blockDim.x) {
    float v = (to_f32<T>(x[i]) - mean) * rstd;
    y[i] = from_f32<T>(v * to_f32<T>(weight[i]) + to_f32<T>(bias[i]));
  }
}

template <typename T>
void launch_ln(const void* in, const void* w, const void* b, void* out,
               long long rows, int d, float eps, hipStream_t stream) {
  int block = d >= 2048 ? 512 : 256;
  hipLaunchKernelGGL((layernorm_kernel<T>), dim3((unsigned)rows), dim3(block),
                     0, stream, (const T*)in, (const T*)w, (const T*)b,
                     (T*)out, (int)rows, d, eps);
}

}  // namespace

extern "C" void vfa_layer_norm(const void* in, const void* w, const void* b,
                               void* out, long long rows, int d, float eps,
                               int dtype, hipStream_t stream) {
  switch (dtype) {
    case VFA_F32: launch_ln<float>(in, w, b, out, rows, d, eps, stream); break;
    case VFA_BF16:
      launch_ln<__hip_bfloat16>(in, w, b, out, rows, d, eps, stream); break;
    case VFA_F16: launch_ln<__half>(in, w, b, out, rows, d, eps, stream); break;
  }
}
