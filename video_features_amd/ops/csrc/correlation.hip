// PWC cost-volume correlation for gfx950.
//
// The reference JIT-compiles four CUDA kernels via CuPy (reference
// models/pwc/pwc_src/correlation.py): NCHW->padded-NHWC rearrange + an
// 81-channel updateOutput with a 32-thread block and __shared__ float
// sum[32] — a Volta-era shape.  Re-designed for CDNA4:
//
//  * repack: NCHW -> zero-padded NHWC (pad = max_disp), coalesced writes.
//  * corr_tiled (C <= 64, the high-resolution pyramid levels): one wave64
//    per 8x8 pixel tile; the f1 tile (8x8xC) and the f2 tile (16x16xC,
//    padding-inclusive) are staged in LDS once and reused by all 81
//    displacements — an 81x reuse factor no cache gives reliably; lane =
//    pixel, no cross-lane reduction at all.
//  * corr_wave (C > 64, the tiny deep levels): one wave per pixel, lanes
//    stride channels (coalesced NHWC reads), f1 cached in VGPRs,
//    wave shuffle reduction per displacement.
//
// Output (B, (2d+1)^2, H, W) = channel-MEAN dot products (reference
// semantics).
#include "vfa_common.h"

namespace {

constexpr int MAXD = 4;           // displacement radius (9x9 window)
constexpr int NDISP = 81;

// ---------------------------------------------------------------- repack
template <typename T>
__global__ void repack_kernel(const T* __restrict__ in, T* __restrict__ out,
                              int b, int c, int h, int w) {
  // out: (b, h+2p, w+2p, c), zero border
  const int hp = h + 2 * MAXD, wp = w + 2 * MAXD;
  const long long total = (long long)b * hp * wp * c;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < total; i += (long long)gridDim.x * blockDim.x) {
    const int ci = (int)(i % c);
    long long r = i / c;
    const int xp = (int)(r % wp);
    r /= wp;
    const int yp = (int)(r % hp);
    const int bi = (int)(r / hp);
    const int y = yp - MAXD, x = xp - MAXD;
    T v = from_f32<T>(0.f);
    if (y >= 0 && y < h && x >= 0 && x < w)
      v = in[(((long long)bi * c + ci) * h + y) * w + x];
    out[i] = v;
  }
}

// ------------------------------------------------------------ corr_tiled
template <typename T>
__global__ void corr_tiled_kernel(const T* __restrict__ f1,   // NCHW
                                  const T* __restrict__ f2p,  // padded NHWC
                                  float* __restrict__ out, int b, int c,
                                  int h, int w) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int cp = c + 1;   // +1 element pad breaks the power-of-2 bank stride
  T* s_f1 = reinterpret_cast<T*>(smem);                 // [64][cp]
  T* s_f2 = s_f1 + 64 * cp;                             // [256][cp]
  const int tileX = blockIdx.x * 8, tileY = blockIdx.y * 8;
  const int bi = blockIdx.z;
  const int lane = threadIdx.x;                         // 0..63
  const int tx = lane & 7, ty = lane >> 3;
  const int x = tileX + tx, y = tileY + ty;
  const int wp = w + 2 * MAXD;
  const long long hw = (long long)h * w;

  // stage f1: each lane loads its own pixel's channel vector (NCHW reads,
  // stride hw — hits L2; done once per 81 reuses)
  if (x < w && y < h) {
    const T* src = f1 + (long long)bi * c * hw + (long long)y * w + x;
    for (int ci = 0; ci < c; ++ci) s_f1[lane * cp + ci] = src[(long long)ci * hw];
  }
  // stage f2: 16x16 padded-NHWC patch rooted at (tileY, tileX) in padded
  // coords; 4 pixels per lane, contiguous channel reads
  for (int p = lane; p < 256; p += 64) {
    const int fy = p >> 4, fx = p & 15;
    const int gy = tileY + fy, gx = tileX + fx;   // padded coords
    T* dst = s_f2 + p * cp;
    if (gy < h + 2 * MAXD && gx < wp) {
      const T* src = f2p + (((long long)bi * (h + 2 * MAXD) + gy) * wp + gx) * c;
      for (int ci = 0; ci < c; ++ci) dst[ci] = src[ci];
    }
  }
  __syncthreads();
  if (x >= w || y >= h) return;

  const float inv_c = 1.0f / c;
  const T* my_f1 = s_f1 + lane * cp;
  float* out_b = out + ((long long)bi * NDISP) * hw + (long long)y * w + x;
#pragma unroll 3
  for (int d = 0; d < NDISP; ++d) {
    const int dy = d / 9, dx = d % 9;            // 0..8 == offset -4..+4
    const T* other = s_f2 + ((ty + dy) * 16 + (tx + dx)) * cp;
    float acc = 0.f;
    for (int ci = 0; ci < c; ++ci)
      acc += to_f32<T>(my_f1[ci]) * to_f32<T>(other[ci]);
    out_b[(long long)d * hw] = acc * inv_c;
  }
}

// ------------------------------------------------------------- corr_wave
template <typename T>
__global__ void corr_wave_kernel(const T* __restrict__ f1p,   // padded NHWC
                                 const T* __restrict__ f2p,   // padded NHWC
                                 float* __restrict__ out, int b, int c,
                                 int h, int w) {
  const int wp = w + 2 * MAXD;
  const long long hw = (long long)h * w;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int waves_per_block = blockDim.x >> 6;
  const long long pix_id =
      (long long)blockIdx.x * waves_per_block + wave;
  if (pix_id >= (long long)b * hw) return;
  const int bi = (int)(pix_id / hw);
  const int pix = (int)(pix_id % hw);
  const int y = pix / w, x = pix % w;

  // f1 channel chunk in registers (lane-strided)
  float reg1[4];
  const T* src1 =
      f1p + (((long long)bi * (h + 2 * MAXD) + y + MAXD) * wp + x + MAXD) * c;
  const int nchunk = (c + 63) / 64;
#pragma unroll
  for (int k = 0; k < 4; ++k) {
    const int ci = k * 64 + lane;
    reg1[k] = (k < nchunk && ci < c) ? to_f32<T>(src1[ci]) : 0.f;
  }
  const float inv_c = 1.0f / c;
  float* out_b = out + ((long long)bi * NDISP) * hw + pix;
  for (int d = 0; d < NDISP; ++d) {
    const int dy = d / 9, dx = d % 9;
    const T* src2 =
        f2p + (((long long)bi * (h + 2 * MAXD) + y + dy) * wp + x + dx) * c;
    float acc = 0.f;
#pragma unroll
    for (int k = 0; k < 4; ++k) {
      const int ci = k * 64 + lane;
      if (k < nchunk && ci < c) acc += reg1[k] * to_f32<T>(src2[ci]);
    }
    acc = wave_reduce_sum(acc);
    if (lane == 0) out_b[(long long)d * hw] = acc * inv_c;
  }
}

template <typename T>
void launch_corr(const void* f1, const void* f1p, const void* f2p, void* out,
                 int b, int c, int h, int w, hipStream_t stream) {
  if (c <= 64) {
    dim3 grid((w + 7) / 8, (h + 7) / 8, b);
    size_t lds = (size_t)(64 + 256) * (c + 1) * sizeof(T);
    hipLaunchKernelGGL((corr_tiled_kernel<T>), grid, dim3(64), lds, stream,
                       (const T*)f1, (const T*)f2p, (float*)out, b, c, h, w);
  } else {
    const int waves_per_block = 4;
    long long npix = (long long)b * h * w;
    long long nblocks = (npix + waves_per_block - 1) / waves_per_block;
    hipLaunchKernelGGL((corr_wave_kernel<T>), dim3((unsigned)nblocks),
                       dim3(waves_per_block * 64), 0, stream, (const T*)f1p,
                       (const T*)f2p, (float*)out, b, c, h, w);
  }
}

}  // namespace

extern "C" {

void vfa_corr_repack(const void* in, void* out, int b, int c, int h, int w,
                     int dtype, hipStream_t stream) {
  const int hp = h + 2 * MAXD, wp = w + 2 * MAXD;
  long long total = (long long)b * hp * wp * c;
  int block = 256;
  int grid = (int)min((total + block - 1) / block, (long long)8192);
  switch (dtype) {
    case VFA_F32:
      hipLaunchKernelGGL((repack_kernel<float>), dim3(grid), dim3(block), 0,
                         stream, (const float*)in, (float*)out, b, c, h, w);
      break;
    case VFA_BF16:
      hipLaunchKernelGGL((repack_kernel<__hip_bfloat16>), dim3(grid),
                         dim3(block), 0, stream, (const __hip_bfloat16*)in,
                         (__hip_bfloat16*)out, b, c, h, w);
      break;
    case VFA_F16:
      hipLaunchKernelGGL((repack_kernel<__half>), dim3(grid), dim3(block), 0,
                         stream, (const __half*)in, (__half*)out, b, c, h, w);
      break;
  }
}

// f1: NCHW original; f1p/f2p: padded NHWC (from vfa_corr_repack);
// out: (B, 81, H, W) float32
void vfa_pwc_correlation(const void* f1, const void* f1p, const void* f2p,
                         void* out, int b, int c, int h, int w, int dtype,
                         hipStream_t stream) {
  switch (dtype) {
    case VFA_F32: launch_corr<float>(f1, f1p, f2p, out, b, c, h, w, stream); break;
    case VFA_BF16:
      launch_corr<__hip_bfloat16>(f1, f1p, f2p, out, b, c, h, w, stream); break;
    case VFA_F16:
      launch_corr<__half>(f1, f1p, f2p, out, b, c, h, w, stream); break;
  }
}

}  // extern "C"
