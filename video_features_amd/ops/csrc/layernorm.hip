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

// residual fusion: s = x + res is written once and normalized in the same
// launch (removes the separate eager add kernel + one full re-read)
template <typename T>
__global__ void layernorm_res_kernel(const T* __restrict__ x,
                                     const T* __restrict__ res,
                                     const T* __restrict__ weight,
                                     const T* __restrict__ bias,
                                     T* __restrict__ y, T* __restrict__ s_out,
                                     int rows, int d, float eps) {
  constexpr int VEC = 16 / sizeof(T);
  __shared__ float red[2][8];
  const int row = blockIdx.x;
  if (row >= rows) return;
  const T* xr = x + (long long)row * d;
  const T* rr = res + (long long)row * d;
  T* sr = s_out + (long long)row * d;
  T* yr = y + (long long)row * d;
  const int tid = threadIdx.x;
  const int lane = tid & 63, wave = tid >> 6;
  const int nwaves = blockDim.x >> 6;
  using VecT = __attribute__((ext_vector_type(4))) unsigned;

  float sum = 0.f, sumsq = 0.f;
  const int dvec = d / VEC;
  for (int i = tid; i < dvec; i += blockDim.x) {
    T tx[VEC], tr[VEC];
    *reinterpret_cast<VecT*>(tx) =
        *reinterpret_cast<const VecT*>(xr + (long long)i * VEC);
    *reinterpret_cast<VecT*>(tr) =
        *reinterpret_cast<const VecT*>(rr + (long long)i * VEC);
#pragma unroll
    for (int j = 0; j < VEC; ++j) {
      float v = to_f32<T>(tx[j]) + to_f32<T>(tr[j]);
      tx[j] = from_f32<T>(v);
      sum += v;
      sumsq += v * v;
    }
    *reinterpret_cast<VecT*>(sr + (long long)i * VEC) =
        *reinterpret_cast<VecT*>(tx);
  }
  for (int i = dvec * VEC + tid; i < d; i += blockDim.x) {
    float v = to_f32<T>(xr[i]) + to_f32<T>(rr[i]);
    sr[i] = from_f32<T>(v);
    sum += v;
    sumsq += v * v;
  }
  sum = wave_reduce_sum(sum);
  sumsq = wave_reduce_sum(sumsq);
  if (lane == 0) { red[0][wave] = sum; red[1][wave] = sumsq; }
  __syncthreads();
  if (tid == 0) {
    float s = 0.f, ss = 0.f;
    for (int w2 = 0; w2 < nwaves; ++w2) { s += red[0][w2]; ss += red[1][w2]; }
    float mean = s / d;
    red[0][0] = mean;
    red[1][0] = rsqrtf(ss / d - mean * mean + eps);
  }
  __syncthreads();
  const float mean = red[0][0], rstd = red[1][0];
  for (int i = tid; i < dvec; i += blockDim.x) {
    T tx[VEC], tw[VEC], tb[VEC];
    *reinterpret_cast<VecT*>(tx) =
        *reinterpret_cast<const VecT*>(sr + (long long)i * VEC);
    *reinterpret_cast<VecT*>(tw) =
        *reinterpret_cast<const VecT*>(weight + (long long)i * VEC);
    *reinterpret_cast<VecT*>(tb) =
        *reinterpret_cast<const VecT*>(bias + (long long)i * VEC);
#pragma unroll
    for (int j = 0; j < VEC; ++j) {
      float v = (to_f32<T>(tx[j]) - mean) * rstd;
      tx[j] = from_f32<T>(v * to_f32<T>(tw[j]) + to_f32<T>(tb[j]));
    }
    *reinterpret_cast<VecT*>(yr + (long long)i * VEC) =
        *reinterpret_cast<VecT*>(tx);
  }
  for (int i = dvec * VEC + tid; i < d; i += blockDim.x) {
    float v = (to_f32<T>(sr[i]) - mean) * rstd;
    yr[i] = from_f32<T>(v * to_f32<T>(weight[i]) + to_f32<T>(bias[i]));
  }
}

// wave-per-row variants for the common d<=4096 case: 4 rows per block,
// wave allreduce only — no LDS, no barriers, all lanes loaded (the
// block-per-row versions above leave (256 - d/VEC) lanes idle and cost two
// barriers; measured 57.8us -> bandwidth-bound for (9600, 768) bf16)
template <typename T>
__global__ void layernorm_wave_kernel(const T* __restrict__ in,
                                      const T* __restrict__ weight,
                                      const T* __restrict__ bias,
                                      T* __restrict__ out, long long rows,
                                      int d, float eps) {
  constexpr int VEC = 16 / sizeof(T);
  const long long row = (long long)blockIdx.x * 4 + (threadIdx.x >> 6);
  if (row >= rows) return;
  const int lane = threadIdx.x & 63;
  const T* x = in + row * d;
  T* y = out + row * d;
  using VecT = __attribute__((ext_vector_type(4))) unsigned;
  const int dvec = d / VEC;
  float sum = 0.f, sumsq = 0.f;
  for (int i = lane; i < dvec; i += 64) {
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
  for (int i = dvec * VEC + lane; i < d; i += 64) {
    float v = to_f32<T>(x[i]);
    sum += v;
    sumsq += v * v;
  }
  sum = wave_allreduce_sum(sum);
  sumsq = wave_allreduce_sum(sumsq);
  const float mean = sum / d;
  const float rstd = rsqrtf(sumsq / d - mean * mean + eps);
  for (int i = lane; i < dvec; i += 64) {
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
  for (int i = dvec * VEC + lane; i < d; i += 64) {
    float v = (to_f32<T>(x[i]) - mean) * rstd;
    y[i] = from_f32<T>(v * to_f32<T>(weight[i]) + to_f32<T>(bias[i]));
  }
}

template <typename T>
__global__ void layernorm_res_wave_kernel(const T* __restrict__ x,
                                          const T* __restrict__ res,
                                          const T* __restrict__ weight,
                                          const T* __restrict__ bias,
                                          T* __restrict__ y,
                                          T* __restrict__ s_out,
                                          long long rows, int d, float eps) {
  constexpr int VEC = 16 / sizeof(T);
  const long long row = (long long)blockIdx.x * 4 + (threadIdx.x >> 6);
  if (row >= rows) return;
  const int lane = threadIdx.x & 63;
  const T* xr = x + row * d;
  const T* rr = res + row * d;
  T* sr = s_out + row * d;
  T* yr = y + row * d;
  using VecT = __attribute__((ext_vector_type(4))) unsigned;
  const int dvec = d / VEC;
  float sum = 0.f, sumsq = 0.f;
  for (int i = lane; i < dvec; i += 64) {
    T tx[VEC], tr[VEC];
    *reinterpret_cast<VecT*>(tx) =
        *reinterpret_cast<const VecT*>(xr + (long long)i * VEC);
    *reinterpret_cast<VecT*>(tr) =
        *reinterpret_cast<const VecT*>(rr + (long long)i * VEC);
#pragma unroll
    for (int j = 0; j < VEC; ++j) {
      float v = to_f32<T>(tx[j]) + to_f32<T>(tr[j]);
      tx[j] = from_f32<T>(v);
      sum += v;
      sumsq += v * v;
    }
    *reinterpret_cast<VecT*>(sr + (long long)i * VEC) =
        *reinterpret_cast<VecT*>(tx);
  }
  for (int i = dvec * VEC + lane; i < d; i += 64) {
    float v = to_f32<T>(xr[i]) + to_f32<T>(rr[i]);
    sr[i] = from_f32<T>(v);
    sum += v;
    sumsq += v * v;
  }
  sum = wave_allreduce_sum(sum);
  sumsq = wave_allreduce_sum(sumsq);
  const float mean = sum / d;
  const float rstd = rsqrtf(sumsq / d - mean * mean + eps);
  for (int i = lane; i < dvec; i += 64) {
    T tx[VEC], tw[VEC], tb[VEC];
    *reinterpret_cast<VecT*>(tx) =
        *reinterpret_cast<const VecT*>(sr + (long long)i * VEC);
    *reinterpret_cast<VecT*>(tw) =
        *reinterpret_cast<const VecT*>(weight + (long long)i * VEC);
    *reinterpret_cast<VecT*>(tb) =
        *reinterpret_cast<const VecT*>(bias + (long long)i * VEC);
#pragma unroll
    for (int j = 0; j < VEC; ++j) {
      float v = (to_f32<T>(tx[j]) - mean) * rstd;
      tx[j] = from_f32<T>(v * to_f32<T>(tw[j]) + to_f32<T>(tb[j]));
    }
    *reinterpret_cast<VecT*>(yr + (long long)i * VEC) =
        *reinterpret_cast<VecT*>(tx);
  }
  for (int i = dvec * VEC + lane; i < d; i += 64) {
    float v = (to_f32<T>(sr[i]) - mean) * rstd;
    yr[i] = from_f32<T>(v * to_f32<T>(weight[i]) + to_f32<T>(bias[i]));
  }
}

template <typename T>
void launch_ln(const void* in, const void* w, const void* b, void* out,
               long long rows, int d, float eps, hipStream_t stream) {
  if (d <= 4096) {
    const unsigned grid = (unsigned)((rows + 3) / 4);
    hipLaunchKernelGGL((layernorm_wave_kernel<T>), dim3(grid), dim3(256), 0,
                       stream, (const T*)in, (const T*)w, (const T*)b,
                       (T*)out, rows, d, eps);
    return;
  }
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

extern "C" void vfa_layer_norm_residual(const void* x, const void* res,
                                        const void* w, const void* b,
                                        void* y, void* s_out, long long rows,
                                        int d, float eps, int dtype,
                                        hipStream_t stream) {
  if (d <= 4096) {
    const unsigned grid = (unsigned)((rows + 3) / 4);
    switch (dtype) {
      case VFA_F32:
        hipLaunchKernelGGL((layernorm_res_wave_kernel<float>), dim3(grid),
                           dim3(256), 0, stream, (const float*)x,
                           (const float*)res, (const float*)w,
                           (const float*)b, (float*)y, (float*)s_out, rows,
                           d, eps);
        break;
      case VFA_BF16:
        hipLaunchKernelGGL((layernorm_res_wave_kernel<__hip_bfloat16>),
                           dim3(grid), dim3(256), 0, stream,
                           (const __hip_bfloat16*)x,
                           (const __hip_bfloat16*)res,
                           (const __hip_bfloat16*)w, (const __hip_bfloat16*)b,
                           (__hip_bfloat16*)y, (__hip_bfloat16*)s_out, rows,
                           d, eps);
        break;
      case VFA_F16:
        hipLaunchKernelGGL((layernorm_res_wave_kernel<__half>), dim3(grid),
                           dim3(256), 0, stream, (const __half*)x,
                           (const __half*)res, (const __half*)w,
                           (const __half*)b, (__half*)y, (__half*)s_out,
                           rows, d, eps);
        break;
    }
    return;
  }
  int block = d >= 2048 ? 512 : 256;
  switch (dtype) {
    case VFA_F32:
      hipLaunchKernelGGL((layernorm_res_kernel<float>), dim3((unsigned)rows),
                         dim3(block), 0, stream, (const float*)x,
                         (const float*)res, (const float*)w, (const float*)b,
                         (float*)y, (float*)s_out, (int)rows, d, eps);
      break;
    case VFA_BF16:
      hipLaunchKernelGGL((layernorm_res_kernel<__hip_bfloat16>),
                         dim3((unsigned)rows), dim3(block), 0, stream,
                         (const __hip_bfloat16*)x, (const __hip_bfloat16*)res,
                         (const __hip_bfloat16*)w, (const __hip_bfloat16*)b,
                         (__hip_bfloat16*)y, (__hip_bfloat16*)s_out,
                         (int)rows, d, eps);
      break;
    case VFA_F16:
      hipLaunchKernelGGL((layernorm_res_kernel<__half>), dim3((unsigned)rows),
                         dim3(block), 0, stream, (const __half*)x,
                         (const __half*)res, (const __half*)w,
                         (const __half*)b, (__half*)y, (__half*)s_out,
                         (int)rows, d, eps);
      break;
  }
}
