// Fused InstanceNorm2d (+optional ReLU) for gfx950.
//
// RAFT's fnet uses InstanceNorm after nearly every conv (reference
// raft_src/extractor.py:118-192).  PyTorch lowers InstanceNorm to
// batch_norm with per-(b,c) stats: one statistics kernel + one transform
// kernel + a separate ReLU — three full HBM round-trips.  This kernel does
// the whole thing in one dispatch (stats pass + normalize pass back to
// back; the tile usually stays in L2 between the passes).
//
// No affine (RAFT's InstanceNorm2d(affine=False)); fp32 accumulation.
//
// NCHW: one block per (b, c), threads grid-stride HW (coalesced).
// NHWC: one block per (b, chunk of 64 channels); lane = channel, the
// block's 4 sub-rows stride over pixels so reads stay coalesced.
#include "vfa_common.h"

namespace {

template <typename T, bool RELU>
__global__ void in2d_nchw_kernel(const T* __restrict__ x, T* __restrict__ out,
                                 int hw, float eps) {
  const long long base = (long long)blockIdx.x * hw;
  float s = 0.f, s2 = 0.f;
  for (int i = threadIdx.x; i < hw; i += blockDim.x) {
    const float v = to_f32<T>(x[base + i]);
    s += v;
    s2 += v * v;
  }
  __shared__ float red[2][16];
  s = wave_allreduce_sum(s);
  s2 = wave_allreduce_sum(s2);
  const int wave = threadIdx.x / 64, lane = threadIdx.x % 64;
  const int nwave = blockDim.x / 64;
  if (lane == 0) { red[0][wave] = s; red[1][wave] = s2; }
  __syncthreads();
  if (wave == 0) {
    s = lane < nwave ? red[0][lane] : 0.f;
    s2 = lane < nwave ? red[1][lane] : 0.f;
    s = wave_allreduce_sum(s);
    s2 = wave_allreduce_sum(s2);
    if (lane == 0) { red[0][0] = s; red[1][0] = s2; }
  }
  __syncthreads();
  const float mean = red[0][0] / hw;
  const float var = red[1][0] / hw - mean * mean;
  const float rstd = rsqrtf(var + eps);
  for (int i = threadIdx.x; i < hw; i += blockDim.x) {
    float v = (to_f32<T>(x[base + i]) - mean) * rstd;
    if (RELU) v = fmaxf(v, 0.f);
    out[base + i] = from_f32<T>(v);
  }
}

// NHWC: grid (chunks_per_batch, B); block (64, ROWS). lane = channel within
// the 64-wide chunk, sub-row strides over pixels.
template <typename T, bool RELU, int ROWS>
__global__ void in2d_nhwc_kernel(const T* __restrict__ x, T* __restrict__ out,
                                 int c, int hw, float eps) {
  const int ch = blockIdx.x * 64 + threadIdx.x;  // channel
  const int row = threadIdx.y;
  const long long base = (long long)blockIdx.y * hw * c;
  float s = 0.f, s2 = 0.f;
  if (ch < c) {
    for (int p = row; p < hw; p += ROWS) {
      const float v = to_f32<T>(x[base + (long long)p * c + ch]);
      s += v;
      s2 += v * v;
    }
  }
  // reduce across the ROWS sub-rows for each lane
  __shared__ float red[2][ROWS][64];
  red[0][row][threadIdx.x] = s;
  red[1][row][threadIdx.x] = s2;
  __syncthreads();
  if (row == 0) {
#pragma unroll
    for (int r = 1; r < ROWS; ++r) {
      s += red[0][r][threadIdx.x];
      s2 += red[1][r][threadIdx.x];
    }
    red[0][0][threadIdx.x] = s;
    red[1][0][threadIdx.x] = s2;
  }
  __syncthreads();
  const float mean = red[0][0][threadIdx.x] / hw;
  const float var = red[1][0][threadIdx.x] / hw - mean * mean;
  const float rstd = rsqrtf(var + eps);
  if (ch < c) {
    for (int p = row; p < hw; p += ROWS) {
      const long long i = base + (long long)p * c + ch;
      float v = (to_f32<T>(x[i]) - mean) * rstd;
      if (RELU) v = fmaxf(v, 0.f);
      out[i] = from_f32<T>(v);
    }
  }
}

template <typename T>
void launch_in2d(const void* x, void* out, int b, int c, int hw, float eps,
                 int relu, int nhwc, hipStream_t stream) {
  if (nhwc) {
    // 16 pixel sub-rows per block: enough waves in flight to hide the
    // strided-load latency of the two passes (941us -> bandwidth-bound
    // at ROWS=4 this kernel was 3x off the HBM floor)
    const dim3 grid((c + 63) / 64, b);
    const dim3 block(64, 16);
    if (relu)
      hipLaunchKernelGGL((in2d_nhwc_kernel<T, true, 16>), grid, block, 0,
                         stream, (const T*)x, (T*)out, c, hw, eps);
    else
      hipLaunchKernelGGL((in2d_nhwc_kernel<T, false, 16>), grid, block, 0,
                         stream, (const T*)x, (T*)out, c, hw, eps);
  } else {
    const dim3 grid(b * c);
    const dim3 block(256);
    if (relu)
      hipLaunchKernelGGL((in2d_nchw_kernel<T, true>), grid, block, 0, stream,
                         (const T*)x, (T*)out, hw, eps);
    else
      hipLaunchKernelGGL((in2d_nchw_kernel<T, false>), grid, block, 0, stream,
                         (const T*)x, (T*)out, hw, eps);
  }
}

}  // namespace

extern "C" {

void vfa_instance_norm2d(const void* x, void* out, int b, int c, int hw,
                         float eps, int relu, int nhwc, int dtype,
                         hipStream_t stream) {
  switch (dtype) {
    case VFA_F32:
      launch_in2d<float>(x, out, b, c, hw, eps, relu, nhwc, stream);
      break;
    case VFA_BF16:
      launch_in2d<__hip_bfloat16>(x, out, b, c, hw, eps, relu, nhwc, stream);
      break;
    case VFA_F16:
      launch_in2d<__half>(x, out, b, c, hw, eps, relu, nhwc, stream);
      break;
  }
}

}  // extern "C"
