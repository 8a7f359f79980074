// TF-SAME max_pool3d for gfx950 (I3D's MaxPool3dTFPadding, reference
// models/i3d/i3d_src/i3d_net.py:108-120).
//
// The reference (and our CPU path) does F.pad(zeros) + nn.MaxPool3d; torch's
// GPU max_pool3d always computes argmax indices as well.  This kernel folds
// the asymmetric TF-SAME padding into the window bounds (no padded copy)
// and skips the indices — one dispatch, one read of the tensor.
// Zero-padding max semantics match F.pad(0) + maxpool exactly: positions
// outside the input contribute the value 0.
#include "vfa_common.h"

namespace {

template <typename T>
__global__ void maxpool3d_same_kernel(const T* __restrict__ x,
                                      T* __restrict__ out, long long bc,
                                      int it, int ih, int iw, int ot, int oh,
                                      int ow, int kt, int kh, int kw, int st,
                                      int sh, int sw, int pt, int ph,
                                      int pw) {
  const long long ohw = (long long)ot * oh * ow;
  const long long total = bc * ohw;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < total; i += (long long)gridDim.x * blockDim.x) {
    const long long n = i / ohw;
    const int rem = (int)(i % ohw);
    const int to = rem / (oh * ow);
    const int yo = (rem / ow) % oh;
    const int xo = rem % ow;
    const int t0 = to * st - pt, y0 = yo * sh - ph, x0 = xo * sw - pw;
    const T* xb = x + n * (long long)it * ih * iw;
    float m = -INFINITY;
    bool padded = false;
    for (int dt = 0; dt < kt; ++dt) {
      const int t = t0 + dt;
      if (t < 0 || t >= it) { padded = true; continue; }
      for (int dy = 0; dy < kh; ++dy) {
        const int y = y0 + dy;
        if (y < 0 || y >= ih) { padded = true; continue; }
        const T* row = xb + ((long long)t * ih + y) * iw;
        for (int dx = 0; dx < kw; ++dx) {
          const int xx = x0 + dx;
          if (xx < 0 || xx >= iw) { padded = true; continue; }
          m = fmaxf(m, to_f32<T>(row[xx]));
        }
      }
    }
    if (padded) m = fmaxf(m, 0.f);  // zero-padding participates in the max
    out[i] = from_f32<T>(m);
  }
}

// spatial 2D TF-SAME max pool (zero padding), NCHW or NHWC — used by the
// flattened (B*T, C, H, W) I3D path where the temporal max is a separate
// shifted elementwise maximum
template <typename T, bool NHWC>
__global__ void maxpool2d_same_kernel(const T* __restrict__ x,
                                      T* __restrict__ out, long long n,
                                      int c, int ih, int iw, int oh, int ow,
                                      int kh, int kw, int sh, int sw, int ph,
                                      int pw) {
  const long long ohwc = (long long)oh * ow * c;
  const long long total = n * ohwc;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < total; i += (long long)gridDim.x * blockDim.x) {
    long long ni;
    int yo, xo, ci;
    if (NHWC) {
      ni = i / ohwc;
      const long long r = i % ohwc;
      yo = (int)(r / ((long long)ow * c));
      xo = (int)((r / c) % ow);
      ci = (int)(r % c);
    } else {
      ni = i / ohwc;
      const long long r = i % ohwc;
      ci = (int)(r / ((long long)oh * ow));
      yo = (int)((r / ow) % oh);
      xo = (int)(r % ow);
    }
    const int y0 = yo * sh - ph, x0 = xo * sw - pw;
    const T* xb = x + ni * (long long)c * ih * iw;
    float m = -INFINITY;
    bool padded = false;
    for (int dy = 0; dy < kh; ++dy) {
      const int y = y0 + dy;
      if (y < 0 || y >= ih) { padded = true; continue; }
      for (int dx = 0; dx < kw; ++dx) {
        const int xx = x0 + dx;
        if (xx < 0 || xx >= iw) { padded = true; continue; }
        const long long xi = NHWC ? (((long long)y * iw + xx) * c + ci)
                                  : (((long long)ci * ih + y) * iw + xx);
        m = fmaxf(m, to_f32<T>(xb[xi]));
      }
    }
    if (padded) m = fmaxf(m, 0.f);
    out[i] = from_f32<T>(m);
  }
}

}  // namespace

extern "C" {

void vfa_maxpool2d_same(const void* x, void* out, long long n, int c, int ih,
                        int iw, int oh, int ow, int kh, int kw, int sh,
                        int sw, int ph, int pw, int nhwc, int dtype,
                        hipStream_t stream) {
  const long long total = n * (long long)c * oh * ow;
  const int grid = (int)min((total + 255) / 256, (long long)65536);
#define VFA_MP2_CASE(T)                                                       \
  if (nhwc)                                                                   \
    hipLaunchKernelGGL((maxpool2d_same_kernel<T, true>), dim3(grid),          \
                       dim3(256), 0, stream, (const T*)x, (T*)out, n, c, ih,  \
                       iw, oh, ow, kh, kw, sh, sw, ph, pw);                   \
  else                                                                        \
    hipLaunchKernelGGL((maxpool2d_same_kernel<T, false>), dim3(grid),         \
                       dim3(256), 0, stream, (const T*)x, (T*)out, n, c, ih,  \
                       iw, oh, ow, kh, kw, sh, sw, ph, pw);
  switch (dtype) {
    case VFA_F32: VFA_MP2_CASE(float) break;
    case VFA_BF16: VFA_MP2_CASE(__hip_bfloat16) break;
    case VFA_F16: VFA_MP2_CASE(__half) break;
  }
#undef VFA_MP2_CASE
}

void vfa_maxpool3d_same(const void* x, void* out, long long bc, int it,
                        int ih, int iw, int ot, int oh, int ow, int kt,
                        int kh, int kw, int st, int sh, int sw, int pt,
                        int ph, int pw, int dtype, hipStream_t stream) {
  const long long total = bc * ot * oh * ow;
  const int grid = (int)min((total + 255) / 256, (long long)16384);
#define VFA_MP_CASE(T)                                                        \
  hipLaunchKernelGGL((maxpool3d_same_kernel<T>), dim3(grid), dim3(256), 0,    \
                     stream, (const T*)x, (T*)out, bc, it, ih, iw, ot, oh,    \
                     ow, kt, kh, kw, st, sh, sw, pt, ph, pw);
  switch (dtype) {
    case VFA_F32: VFA_MP_CASE(float) break;
    case VFA_BF16: VFA_MP_CASE(__hip_bfloat16) break;
    case VFA_F16: VFA_MP_CASE(__half) break;
  }
#undef VFA_MP_CASE
}

}  // extern "C"
