// Fused MHSA core for short sequences (ViT frames: N<=64 tokens, D<=128)
// on gfx950.  One workgroup per (batch, head): K and V staged once in LDS
// with a +1-dword row pad (conflict-free column reads, guide G4), then each
// wave processes query rows round-robin:
//   phase 1 — lane j owns key j: s_j = q . K[j] (q via LDS broadcast reads),
//   phase 2 — wave-allreduce softmax (max+sum over the 64 lanes),
//   phase 3 — lane d owns out dim d: o_d = sum_j p_j V[j][d] (p via LDS
//             broadcast), fully coalesced stores.
// No cross-lane reductions in any inner loop; softmax costs 12 shuffles per
// row total.  At ViT-B/32 scale the attention core is a few % of model
// FLOPs (the GEMMs dominate) — this kernel removes the 5-kernel eager
// softmax chain rather than chasing MFMA peak.
#include "vfa_common.h"

namespace {

template <typename T>
__global__ void mhsa_small_kernel(const T* __restrict__ q,
                                  const T* __restrict__ k,
                                  const T* __restrict__ v,
                                  T* __restrict__ out, int bh, int n, int d,
                                  float scale) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int stride = d + 2;                       // elements; +2 keeps 4B pad
  T* s_k = reinterpret_cast<T*>(smem);            // [n][stride]
  T* s_v = s_k + 64 * stride;                     // [n][stride]
  float* s_q = reinterpret_cast<float*>(s_v + 64 * stride);   // [4][128]
  float* s_p = s_q + 4 * 128;                     // [4][64]

  const int bhi = blockIdx.x;
  if (bhi >= bh) return;
  const long long base = (long long)bhi * n * d;
  const int tid = threadIdx.x, lane = tid & 63, wave = tid >> 6;
  const int nwaves = blockDim.x >> 6;

  // stage K and V (row per lane set, contiguous d reads)
  for (int row = tid; row < n; row += blockDim.x) {
    const T* ks = k + base + (long long)row * d;
    const T* vs = v + base + (long long)row * d;
    T* kd = s_k + row * stride;
    T* vd = s_v + row * stride;
    for (int di = 0; di < d; ++di) { kd[di] = ks[di]; vd[di] = vs[di]; }
  }
  __syncthreads();

  for (int row = wave; row < n; row += nwaves) {
    // q row into LDS as f32 (broadcast-friendly)
    const T* qs = q + base + (long long)row * d;
    for (int di = lane; di < d; di += 64)
      s_q[wave * 128 + di] = to_f32<T>(qs[di]) * scale;
    // cross-LANE LDS hand-off within one wave: drain the LDS queue and pin
    // program order before other lanes read what this lane wrote (no
    // __syncthreads here — waves run different row counts)
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    float s = -1e30f;
    if (lane < n) {
      const T* krow = s_k + lane * stride;
      float acc = 0.f;
      for (int di = 0; di < d; ++di)
        acc += s_q[wave * 128 + di] * to_f32<T>(krow[di]);
      s = acc;
    }
    const float m = wave_allreduce_max(s);
    float p = (lane < n) ? __expf(s - m) : 0.f;
    const float denom = wave_allreduce_sum(p);
    s_p[wave * 64 + lane] = p / denom;
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    T* orow = out + base + (long long)row * d;
    for (int di = lane; di < d; di += 64) {
      float acc = 0.f;
      for (int j = 0; j < n; ++j)
        acc += s_p[wave * 64 + j] * to_f32<T>(s_v[j * stride + di]);
      orow[di] = from_f32<T>(acc);
    }
  }
}

template <typename T>
void launch_mhsa(const void* q, const void* k, const void* v, void* out,
                 int bh, int n, int d, float scale, hipStream_t stream) {
  const int stride = d + 2;
  size_t lds = (size_t)2 * 64 * stride * sizeof(T) +
               (4 * 128 + 4 * 64) * sizeof(float);
  hipLaunchKernelGGL((mhsa_small_kernel<T>), dim3(bh), dim3(256), lds, stream,
                     (const T*)q, (const T*)k, (const T*)v, (T*)out, bh, n, d,
                     scale);
}

}  // namespace

extern "C" void vfa_mhsa_small(const void* q, const void* k, const void* v,
                               void* out, int bh, int n, int d, float scale,
                               int dtype, hipStream_t stream) {
  switch (dtype) {
    case VFA_F32: launch_mhsa<float>(q, k, v, out, bh, n, d, scale, stream); break;
    case VFA_BF16:
      launch_mhsa<__hip_bfloat16>(q, k, v, out, bh, n, d, scale, stream); break;
    case VFA_F16:
      launch_mhsa<__half>(q, k, v, out, bh, n, d, scale, stream); break;
  }
}
