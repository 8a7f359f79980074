// Bilinear gather kernels for optical flow on gfx950:
//  - vfa_bilinear_warp: backward warp x (B,C,H,W) by flow (B,2,H,W) with
//    border zero-masking (PWC `Backward`, reference pwc_net.py:23-41)
//  - vfa_grid_sample: RAFT pyramid lookup — pixel-unit coords
//    (N,Ho,Wo,2), zeros padding, align_corners=True semantics
//    (reference raft_src/utils/utils.py:57-71)
#include "vfa_common.h"

namespace {

template <typename T>
__global__ void bilinear_warp_kernel(const T* __restrict__ x,
                                     const T* __restrict__ flow,
                                     T* __restrict__ out, int b, int c, int h,
                                     int w) {
  const long long hw = (long long)h * w;
  const long long total = (long long)b * hw;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < total; i += (long long)gridDim.x * blockDim.x) {
    const int pix = (int)(i % hw);
    const int bi = (int)(i / hw);
    const int yy = pix / w, xx = pix % w;
    const float fx = to_f32<T>(flow[((long long)bi * 2 + 0) * hw + pix]);
    const float fy = to_f32<T>(flow[((long long)bi * 2 + 1) * hw + pix]);
    const float sx = xx + fx, sy = yy + fy;
    const int x0 = (int)floorf(sx), y0 = (int)floorf(sy);
    const float ax = sx - x0, ay = sy - y0;
    // mask semantics: fully-inside support only (grid_sample zeros + >0.999
    // mask in the torch reference path)
    const bool inside =
        (sx >= 0.f) && (sy >= 0.f) && (sx <= w - 1.f) && (sy <= h - 1.f);
    const int x1 = min(x0 + 1, w - 1), y1 = min(y0 + 1, h - 1);
    const int cx0 = max(x0, 0), cy0 = max(y0, 0);
    const float w00 = (1 - ax) * (1 - ay), w01 = ax * (1 - ay);
    const float w10 = (1 - ax) * ay, w11 = ax * ay;
    const T* xb = x + (long long)bi * c * hw;
    T* ob = out + (long long)bi * c * hw;
    for (int ci = 0; ci < c; ++ci) {
      float v = 0.f;
      if (inside) {
        const T* xc = xb + (long long)ci * hw;
        v = w00 * to_f32<T>(xc[cy0 * w + cx0]) +
            w01 * to_f32<T>(xc[cy0 * w + x1]) +
            w10 * to_f32<T>(xc[y1 * w + cx0]) +
            w11 * to_f32<T>(xc[y1 * w + x1]);
      }
      ob[(long long)ci * hw + pix] = from_f32<T>(v);
    }
  }
}

template <typename T>
__global__ void grid_sample_kernel(const T* __restrict__ x,
                                   const T* __restrict__ coords,
                                   T* __restrict__ out, long long n, int c,
                                   int h, int w, int ho, int wo) {
  const long long owh = (long long)ho * wo;
  const long long total = n * owh;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < total; i += (long long)gridDim.x * blockDim.x) {
    const long long ni = i / owh;
    const int opix = (int)(i % owh);
    const float sx = to_f32<T>(coords[i * 2 + 0]);
    const float sy = to_f32<T>(coords[i * 2 + 1]);
    const int x0 = (int)floorf(sx), y0 = (int)floorf(sy);
    const float ax = sx - x0, ay = sy - y0;
    const float w00 = (1 - ax) * (1 - ay), w01 = ax * (1 - ay);
    const float w10 = (1 - ax) * ay, w11 = ax * ay;
    const bool i00 = (x0 >= 0) & (x0 < w) & (y0 >= 0) & (y0 < h);
    const bool i01 = (x0 + 1 >= 0) & (x0 + 1 < w) & (y0 >= 0) & (y0 < h);
    const bool i10 = (x0 >= 0) & (x0 < w) & (y0 + 1 >= 0) & (y0 + 1 < h);
    const bool i11 = (x0 + 1 >= 0) & (x0 + 1 < w) & (y0 + 1 >= 0) & (y0 + 1 < h);
    const long long hw = (long long)h * w;
    const T* xb = x + ni * c * hw;
    T* ob = out + ni * c * owh;
    for (int ci = 0; ci < c; ++ci) {
      const T* xc = xb + (long long)ci * hw;
      float v = 0.f;
      if (i00) v += w00 * to_f32<T>(xc[(long long)y0 * w + x0]);
      if (i01) v += w01 * to_f32<T>(xc[(long long)y0 * w + x0 + 1]);
      if (i10) v += w10 * to_f32<T>(xc[(long long)(y0 + 1) * w + x0]);
      if (i11) v += w11 * to_f32<T>(xc[(long long)(y0 + 1) * w + x0 + 1]);
      ob[(long long)ci * owh + opix] = from_f32<T>(v);
    }
  }
}

template <typename T>
void launch_warp(const void* x, const void* flow, void* out, int b, int c,
                 int h, int w, hipStream_t stream) {
  long long total = (long long)b * h * w;
  int block = 256;
  int grid = (int)min((total + block - 1) / block, (long long)4096);
  hipLaunchKernelGGL((bilinear_warp_kernel<T>), dim3(grid), dim3(block), 0,
                     stream, (const T*)x, (const T*)flow, (T*)out, b, c, h, w);
}

template <typename T>
void launch_gs(const void* x, const void* coords, void* out, long long n,
               int c, int h, int w, int ho, int wo, hipStream_t stream) {
  long long total = n * ho * wo;
  int block = 256;
  int grid = (int)min((total + block - 1) / block, (long long)8192);
  hipLaunchKernelGGL((grid_sample_kernel<T>), dim3(grid), dim3(block), 0,
                     stream, (const T*)x, (const T*)coords, (T*)out, n, c, h,
                     w, ho, wo);
}

}  // namespace

extern "C" {

void vfa_bilinear_warp(const void* x, const void* flow, void* out, int b,
                       int c, int h, int w, int dtype, hipStream_t stream) {
  switch (dtype) {
    case VFA_F32: launch_warp<float>(x, flow, out, b, c, h, w, stream); break;
    case VFA_BF16:
      launch_warp<__hip_bfloat16>(x, flow, out, b, c, h, w, stream); break;
    case VFA_F16: launch_warp<__half>(x, flow, out, b, c, h, w, stream); break;
  }
}

void vfa_grid_sample(const void* x, const void* coords, void* out,
                     long long n, int c, int h, int w, int ho, int wo,
                     int dtype, hipStream_t stream) {
  switch (dtype) {
    case VFA_F32: launch_gs<float>(x, coords, out, n, c, h, w, ho, wo, stream); break;
    case VFA_BF16:
      launch_gs<__hip_bfloat16>(x, coords, out, n, c, h, w, ho, wo, stream); break;
    case VFA_F16:
      launch_gs<__half>(x, coords, out, n, c, h, w, ho, wo, stream); break;
  }
}

}  // extern "C"
