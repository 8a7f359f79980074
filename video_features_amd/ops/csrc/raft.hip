// RAFT-specific fused kernels for gfx950 (CDNA4).
//
// The reference RAFT (models/raft/raft_src/{corr,raft,update}.py) runs its
// 20-iteration update loop as dozens of small torch ops per iteration:
// 4 grid_sample lookups + window-coordinate construction + cat/permute for
// the correlation pyramid (corr.py:36-50), F.unfold + softmax + mul-sum for
// the convex 8x upsample (raft.py:100-111), and 6 convs + 6 elementwise for
// the SepConvGRU (update.py:37-64).  On MI355X that is launch-bound; these
// kernels fuse each group into one dispatch:
//
//  - vfa_corr_lookup: all 4 pyramid levels x 81 taps in one kernel, the
//    per-pixel correlation planes staged through LDS (each output pixel owns
//    a whole (h_l, w_l) plane per level; total ~4.2 KB for 28x28 feature
//    maps, far under the 160 KB LDS/CU).  Zero-padding bilinear semantics
//    identical to torch grid_sample(align_corners=True, padding='zeros').
//  - vfa_convex_upsample: softmax(0.25*mask) convex combination of the 3x3
//    neighborhood of 8*flow, one thread per upsampled pixel.
//  - vfa_gru_zr / vfa_gru_out: the SepConvGRU gate elementwise, fused so the
//    z,r sigmoid + r*h product and the (1-z)*h + z*tanh(q) update are one
//    kernel each, writing r*h / h_new directly into the channel slice of the
//    persistent conv-input buffers (no cat per step).
//
// All kernels take an `nhwc` flag: the RAFT GPU path runs channels_last so
// MIOpen uses its NHWC igemm solvers without batched_transpose fixups.
#include "vfa_common.h"

namespace {

// ------------------------------------------------------------ corr lookup
// Pyramid: LV levels, level l is (N, 1, h[l], w[l]) fp32, N = B*H*W planes
// (one correlation plane per query pixel).  coords: (B, 2, H, W) fp32 pixel
// units at level 0.  out: (B, LV*81, H, W) or NHWC (B, H, W, LV*81) in T.
// PPB query pixels per block (one 64-lane wave each): 16 waves in flight
// per block hide the LDS-staging load latency
template <typename T, int RAD, bool NHWC, int PPB>
__global__ void corr_lookup_lds_kernel(const float* __restrict__ l0,
                                       const float* __restrict__ l1,
                                       const float* __restrict__ l2,
                                       const float* __restrict__ l3,
                                       const float* __restrict__ coords,
                                       T* __restrict__ out, int levels,
                                       int4 hs, int4 ws, long long npix,
                                       int hw, int lds_floats) {
  extern __shared__ float lds_all[];
  const int K = 2 * RAD + 1;          // 9
  const int KK = K * K;               // 81 taps per level
  const long long pix = blockIdx.x * (long long)PPB + threadIdx.y;
  if (pix >= npix) return;
  float* lds = lds_all + (long long)threadIdx.y * lds_floats;
  const int tid = threadIdx.x;

  const int lh[4] = {hs.x, hs.y, hs.z, hs.w};
  const int lw[4] = {ws.x, ws.y, ws.z, ws.w};
  const float* lp[4] = {l0, l1, l2, l3};

  // cooperative stage of this pixel's planes (all levels) into LDS
  int off = 0;
  int loff[4];
  for (int l = 0; l < levels; ++l) {
    const int n = lh[l] * lw[l];
    const float* src = lp[l] + pix * (long long)n;
    loff[l] = off;
    for (int i = tid; i < n; i += 64) lds[off + i] = src[i];
    off += n;
  }
  __builtin_amdgcn_wave_barrier();  // each wave stages its own pixel's planes

  const long long b = pix / hw;
  const int p = (int)(pix % hw);
  const float cx = coords[(b * 2 + 0) * hw + p];
  const float cy = coords[(b * 2 + 1) * hw + p];

  const int ctot = levels * KK;
  for (int c = tid; c < ctot; c += 64) {
    const int l = c / KK, t = c % KK;
    const int dx = t / K - RAD, dy = t % K - RAD;  // tap t = i*9+j: i
    // offsets X, j offsets Y — the reference's channel order
    // (corr.py:39 stacks (dy,dx) last, added to (x,y) coords)
    const float inv = 1.0f / (float)(1 << l);
    const float sx = cx * inv + dx, sy = cy * inv + dy;
    const int h = lh[l], w = lw[l];
    const int x0 = (int)floorf(sx), y0 = (int)floorf(sy);
    const float ax = sx - x0, ay = sy - y0;
    const float w00 = (1 - ax) * (1 - ay), w01 = ax * (1 - ay);
    const float w10 = (1 - ax) * ay, w11 = ax * ay;
    const float* pl = lds + loff[l];
    float v = 0.f;
    if (x0 >= 0 && x0 < w && y0 >= 0 && y0 < h) v += w00 * pl[y0 * w + x0];
    if (x0 + 1 >= 0 && x0 + 1 < w && y0 >= 0 && y0 < h)
      v += w01 * pl[y0 * w + x0 + 1];
    if (x0 >= 0 && x0 < w && y0 + 1 >= 0 && y0 + 1 < h)
      v += w10 * pl[(y0 + 1) * w + x0];
    if (x0 + 1 >= 0 && x0 + 1 < w && y0 + 1 >= 0 && y0 + 1 < h)
      v += w11 * pl[(y0 + 1) * w + x0 + 1];
    const long long oi = NHWC ? (pix * ctot + c)
                              : ((b * ctot + c) * hw + p);
    out[oi] = from_f32<T>(v);
  }
}

// global-memory fallback for feature maps too large to stage in LDS
template <typename T, int RAD, bool NHWC>
__global__ void corr_lookup_gmem_kernel(const float* __restrict__ l0,
                                        const float* __restrict__ l1,
                                        const float* __restrict__ l2,
                                        const float* __restrict__ l3,
                                        const float* __restrict__ coords,
                                        T* __restrict__ out, int levels,
                                        int4 hs, int4 ws, long long npix,
                                        int hw) {
  const int K = 2 * RAD + 1, KK = K * K;
  const int ctot = levels * KK;
  const long long total = npix * ctot;
  const int lh[4] = {hs.x, hs.y, hs.z, hs.w};
  const int lw[4] = {ws.x, ws.y, ws.z, ws.w};
  const float* lp[4] = {l0, l1, l2, l3};
  long long loffs[4];
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < total; i += (long long)gridDim.x * blockDim.x) {
    const long long pix = i / ctot;
    const int c = (int)(i % ctot);
    const long long b = pix / hw;
    const int p = (int)(pix % hw);
    const float cx = coords[(b * 2 + 0) * hw + p];
    const float cy = coords[(b * 2 + 1) * hw + p];
    const int l = c / KK, t = c % KK;
    const int dx = t / K - RAD, dy = t % K - RAD;  // tap t = i*9+j: i
    // offsets X, j offsets Y — the reference's channel order
    // (corr.py:39 stacks (dy,dx) last, added to (x,y) coords)
    const float inv = 1.0f / (float)(1 << l);
    const float sx = cx * inv + dx, sy = cy * inv + dy;
    const int h = lh[l], w = lw[l];
    const float* pl = lp[l] + pix * (long long)(h * w);
    const int x0 = (int)floorf(sx), y0 = (int)floorf(sy);
    const float ax = sx - x0, ay = sy - y0;
    const float w00 = (1 - ax) * (1 - ay), w01 = ax * (1 - ay);
    const float w10 = (1 - ax) * ay, w11 = ax * ay;
    float v = 0.f;
    if (x0 >= 0 && x0 < w && y0 >= 0 && y0 < h) v += w00 * pl[y0 * w + x0];
    if (x0 + 1 >= 0 && x0 + 1 < w && y0 >= 0 && y0 < h)
      v += w01 * pl[y0 * w + x0 + 1];
    if (x0 >= 0 && x0 < w && y0 + 1 >= 0 && y0 + 1 < h)
      v += w10 * pl[(y0 + 1) * w + x0];
    if (x0 + 1 >= 0 && x0 + 1 < w && y0 + 1 >= 0 && y0 + 1 < h)
      v += w11 * pl[(y0 + 1) * w + x0 + 1];
    const long long oi = NHWC ? (pix * ctot + c)
                              : ((b * ctot + c) * hw + p);
    out[oi] = from_f32<T>(v);
  }
  (void)loffs;
}

// -------------------------------------------------------- convex upsample
// flow: (B,2,h,w); mask: (B,576,h,w) raw conv output (the 0.25 scale of
// reference raft.py:158 is folded in here); out: (B,2,8h,8w) NCHW always.
// Each thread handles one upsampled pixel (both channels): softmax over the
// 9 mask logits, convex-combine the 3x3 neighborhood of 8*flow.
template <typename T, bool NHWC>
__global__ void convex_upsample_kernel(const T* __restrict__ flow,
                                       const T* __restrict__ mask,
                                       T* __restrict__ out, int b, int h,
                                       int w) {
  const long long hw = (long long)h * w;
  const long long total = (long long)b * hw * 64;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < total; i += (long long)gridDim.x * blockDim.x) {
    const int f = (int)(i % 64);           // fy*8+fx
    const long long pix = i / 64;          // b*hw + ly*w + lx
    const long long bi = pix / hw;
    const int p = (int)(pix % hw);
    const int ly = p / w, lx = p % w;
    // 9 mask logits for this upsampled position
    float m[9], mmax = -1e30f;
#pragma unroll
    for (int k = 0; k < 9; ++k) {
      const int ch = k * 64 + f;
      const long long mi = NHWC ? (pix * 576 + ch) : ((bi * 576 + ch) * hw + p);
      m[k] = 0.25f * to_f32<T>(mask[mi]);
      mmax = fmaxf(mmax, m[k]);
    }
    float msum = 0.f;
#pragma unroll
    for (int k = 0; k < 9; ++k) {
      m[k] = __expf(m[k] - mmax);
      msum += m[k];
    }
    const float rs = 1.0f / msum;
    float acc0 = 0.f, acc1 = 0.f;
#pragma unroll
    for (int k = 0; k < 9; ++k) {
      const int yy = ly + k / 3 - 1, xx = lx + k % 3 - 1;
      if (yy < 0 || yy >= h || xx < 0 || xx >= w) continue;
      const float wk = m[k] * rs;
      const long long fp = (long long)yy * w + xx;
      const long long f0 = NHWC ? ((bi * hw + fp) * 2 + 0)
                                : ((bi * 2 + 0) * hw + fp);
      const long long f1 = NHWC ? ((bi * hw + fp) * 2 + 1)
                                : ((bi * 2 + 1) * hw + fp);
      acc0 += wk * to_f32<T>(flow[f0]);
      acc1 += wk * to_f32<T>(flow[f1]);
    }
    const int oy = ly * 8 + f / 8, ox = lx * 8 + f % 8;
    const long long ohw = hw * 64;
    const long long op = (long long)oy * (8 * w) + ox;
    out[(bi * 2 + 0) * ohw + op] = from_f32<T>(8.f * acc0);
    out[(bi * 2 + 1) * ohw + op] = from_f32<T>(8.f * acc1);
  }
}

// ------------------------------------------------------------- GRU gates
// zr: (B, 2C, h, w) conv output; hx/rhx: persistent (B, C+X, h, w) buffers
// whose first C channels hold h / r*h; z: (B, C, h, w) scratch.
//   z = sigmoid(zr[:, :C]);  r = sigmoid(zr[:, C:2C]);  rhx[:, :C] = r * h
template <typename T, bool NHWC>
__global__ void gru_zr_kernel(const T* __restrict__ zr,
                              const T* __restrict__ hx, T* __restrict__ rhx,
                              T* __restrict__ z, int b, int c, int cx,
                              long long hw) {
  const long long total = (long long)b * c * hw;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < total; i += (long long)gridDim.x * blockDim.x) {
    long long bi, zi0, zi1, si;  // batch, z-idx, r-idx, slice idx (h / rh)
    int ci;
    long long p;
    if (NHWC) {
      const long long pixc = i;            // b*hw*c ordering: pix-major
      const long long pix = pixc / c;
      ci = (int)(pixc % c);
      bi = pix / hw;
      p = pix % hw;
      zi0 = pix * (2 * c) + ci;
      zi1 = zi0 + c;
      si = pix * (c + cx) + ci;
    } else {
      bi = i / (c * hw);
      const long long r = i % (c * hw);
      ci = (int)(r / hw);
      p = r % hw;
      zi0 = (bi * 2 * c + ci) * hw + p;
      zi1 = zi0 + (long long)c * hw;
      si = (bi * (c + cx) + ci) * hw + p;
    }
    const float zv = 1.0f / (1.0f + __expf(-to_f32<T>(zr[zi0])));
    const float rv = 1.0f / (1.0f + __expf(-to_f32<T>(zr[zi1])));
    z[i] = from_f32<T>(zv);
    rhx[si] = from_f32<T>(rv * to_f32<T>(hx[si]));
  }
}

//   h_new = (1-z)*h + z*tanh(q), written in place into hx[:, :C]
template <typename T, bool NHWC>
__global__ void gru_out_kernel(const T* __restrict__ q,
                               const T* __restrict__ z, T* __restrict__ hx,
                               int b, int c, int cx, long long hw) {
  const long long total = (long long)b * c * hw;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < total; i += (long long)gridDim.x * blockDim.x) {
    long long si;
    if (NHWC) {
      const long long pix = i / c;
      const int ci = (int)(i % c);
      si = pix * (c + cx) + ci;
    } else {
      const long long bi = i / (c * hw);
      const long long r = i % (c * hw);
      si = (bi * (c + cx) + r / hw) * hw + r % hw;
    }
    const float zv = to_f32<T>(z[i]);
    const float qv = tanhf(to_f32<T>(q[i]));
    hx[si] = from_f32<T>((1.0f - zv) * to_f32<T>(hx[si]) + zv * qv);
  }
}

template <typename T>
void launch_corr_lookup(const float* l0, const float* l1, const float* l2,
                        const float* l3, const float* coords, void* out,
                        int levels, int4 hs, int4 ws, long long npix, int hw,
                        int lds_floats, int nhwc, hipStream_t stream) {
  constexpr int PPB = 8;
  if (lds_floats > 0 && lds_floats * 4 * PPB <= 64 * 1024) {
    const dim3 grid((unsigned)((npix + PPB - 1) / PPB));
    const dim3 block(64, PPB);
    if (nhwc)
      hipLaunchKernelGGL((corr_lookup_lds_kernel<T, 4, true, PPB>), grid,
                         block, lds_floats * 4 * PPB, stream, l0, l1, l2, l3,
                         coords, (T*)out, levels, hs, ws, npix, hw,
                         lds_floats);
    else
      hipLaunchKernelGGL((corr_lookup_lds_kernel<T, 4, false, PPB>), grid,
                         block, lds_floats * 4 * PPB, stream, l0, l1, l2, l3,
                         coords, (T*)out, levels, hs, ws, npix, hw,
                         lds_floats);
  } else {
    const long long total = npix * levels * 81;
    const int grid = (int)min((total + 255) / 256, (long long)16384);
    if (nhwc)
      hipLaunchKernelGGL((corr_lookup_gmem_kernel<T, 4, true>), dim3(grid),
                         dim3(256), 0, stream, l0, l1, l2, l3, coords,
                         (T*)out, levels, hs, ws, npix, hw);
    else
      hipLaunchKernelGGL((corr_lookup_gmem_kernel<T, 4, false>), dim3(grid),
                         dim3(256), 0, stream, l0, l1, l2, l3, coords,
                         (T*)out, levels, hs, ws, npix, hw);
  }
}

}  // namespace

extern "C" {

void vfa_corr_lookup(const void* l0, const void* l1, const void* l2,
                     const void* l3, const void* coords, void* out,
                     int levels, int h0, int w0, int h1, int w1, int h2,
                     int w2, int h3, int w3, long long npix, int hw,
                     int lds_floats, int nhwc, int dtype,
                     hipStream_t stream) {
  const int4 hs = {h0, h1, h2, h3};
  const int4 ws = {w0, w1, w2, w3};
  switch (dtype) {
    case VFA_F32:
      launch_corr_lookup<float>((const float*)l0, (const float*)l1,
                                (const float*)l2, (const float*)l3,
                                (const float*)coords, out, levels, hs, ws,
                                npix, hw, lds_floats, nhwc, stream);
      break;
    case VFA_BF16:
      launch_corr_lookup<__hip_bfloat16>((const float*)l0, (const float*)l1,
                                         (const float*)l2, (const float*)l3,
                                         (const float*)coords, out, levels,
                                         hs, ws, npix, hw, lds_floats, nhwc,
                                         stream);
      break;
    case VFA_F16:
      launch_corr_lookup<__half>((const float*)l0, (const float*)l1,
                                 (const float*)l2, (const float*)l3,
                                 (const float*)coords, out, levels, hs, ws,
                                 npix, hw, lds_floats, nhwc, stream);
      break;
  }
}

void vfa_convex_upsample(const void* flow, const void* mask, void* out, int b,
                         int h, int w, int nhwc, int dtype,
                         hipStream_t stream) {
  const long long total = (long long)b * h * w * 64;
  const int grid = (int)min((total + 255) / 256, (long long)16384);
#define VFA_CU_CASE(T)                                                        \
  if (nhwc)                                                                   \
    hipLaunchKernelGGL((convex_upsample_kernel<T, true>), dim3(grid),         \
                       dim3(256), 0, stream, (const T*)flow, (const T*)mask,  \
                       (T*)out, b, h, w);                                     \
  else                                                                        \
    hipLaunchKernelGGL((convex_upsample_kernel<T, false>), dim3(grid),        \
                       dim3(256), 0, stream, (const T*)flow, (const T*)mask,  \
                       (T*)out, b, h, w);
  switch (dtype) {
    case VFA_F32: VFA_CU_CASE(float) break;
    case VFA_BF16: VFA_CU_CASE(__hip_bfloat16) break;
    case VFA_F16: VFA_CU_CASE(__half) break;
  }
#undef VFA_CU_CASE
}

void vfa_gru_zr(const void* zr, const void* hx, void* rhx, void* z, int b,
                int c, int cx, long long hw, int nhwc, int dtype,
                hipStream_t stream) {
  const long long total = (long long)b * c * hw;
  const int grid = (int)min((total + 255) / 256, (long long)16384);
#define VFA_ZR_CASE(T)                                                        \
  if (nhwc)                                                                   \
    hipLaunchKernelGGL((gru_zr_kernel<T, true>), dim3(grid), dim3(256), 0,    \
                       stream, (const T*)zr, (const T*)hx, (T*)rhx, (T*)z, b, \
                       c, cx, hw);                                            \
  else                                                                        \
    hipLaunchKernelGGL((gru_zr_kernel<T, false>), dim3(grid), dim3(256), 0,   \
                       stream, (const T*)zr, (const T*)hx, (T*)rhx, (T*)z, b, \
                       c, cx, hw);
  switch (dtype) {
    case VFA_F32: VFA_ZR_CASE(float) break;
    case VFA_BF16: VFA_ZR_CASE(__hip_bfloat16) break;
    case VFA_F16: VFA_ZR_CASE(__half) break;
  }
#undef VFA_ZR_CASE
}

void vfa_gru_out(const void* q, const void* z, void* hx, int b, int c, int cx,
                 long long hw, int nhwc, int dtype, hipStream_t stream) {
  const long long total = (long long)b * c * hw;
  const int grid = (int)min((total + 255) / 256, (long long)16384);
#define VFA_GO_CASE(T)                                                        \
  if (nhwc)                                                                   \
    hipLaunchKernelGGL((gru_out_kernel<T, true>), dim3(grid), dim3(256), 0,   \
                       stream, (const T*)q, (const T*)z, (T*)hx, b, c, cx,    \
                       hw);                                                   \
  else                                                                        \
    hipLaunchKernelGGL((gru_out_kernel<T, false>), dim3(grid), dim3(256), 0,  \
                       stream, (const T*)q, (const T*)z, (T*)hx, b, c, cx,    \
                       hw);
  switch (dtype) {
    case VFA_F32: VFA_GO_CASE(float) break;
    case VFA_BF16: VFA_GO_CASE(__hip_bfloat16) break;
    case VFA_F16: VFA_GO_CASE(__half) break;
  }
#undef VFA_GO_CASE
}

}  // extern "C"
