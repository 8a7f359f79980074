// Fused temporal merge for the flattened-time 3D-conv decomposition
// (models/_flat3d.py): the kt temporal-tap conv outputs live as channel
// groups of y (B*T, kt*O, H, W) channels_last; the merged result is
//   out[b, j, c, p] = sum_dt y[b, j*st - p0 + dt, dt*O + c, p]
// with out-of-range taps contributing zero (temporal zero padding).
// One gather kernel replaces the strided init-copy + (kt-1) strided adds
// (4 passes -> kt reads + 1 write, fully coalesced in NHWC).
#include "vfa_common.h"

namespace {

template <typename T, bool RELU>
__global__ void temporal_merge_kernel(const T* __restrict__ y,
                                      T* __restrict__ out, int b, int t,
                                      int to, int kt, int st, int p0, int o,
                                      long long hw) {
  const long long per_frame_out = hw * o;
  const long long total = (long long)b * to * per_frame_out;
  const int cin = kt * o;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < total; i += (long long)gridDim.x * blockDim.x) {
    const long long bto = i / per_frame_out;
    const long long rem = i % per_frame_out;
    const long long p = rem / o;
    const int c = (int)(rem % o);
    const int bi = (int)(bto / to), j = (int)(bto % to);
    float acc = 0.f;
    for (int dt = 0; dt < kt; ++dt) {
      const int src = j * st - p0 + dt;
      if (src < 0 || src >= t) continue;
      acc += to_f32<T>(y[(((long long)bi * t + src) * hw + p) * cin +
                         dt * o + c]);
    }
    out[i] = from_f32<T>(RELU ? fmaxf(acc, 0.f) : acc);
  }
}

}  // namespace

extern "C" {

void vfa_temporal_merge(const void* y, void* out, int b, int t, int to,
                        int kt, int st, int p0, int o, long long hw,
                        int relu, int dtype, hipStream_t stream) {
  const long long total = (long long)b * to * hw * o;
  const int grid = (int)min((total + 255) / 256, (long long)65536);
#define VFA_TM_CASE(T)                                                        \
  if (relu)                                                                   \
    hipLaunchKernelGGL((temporal_merge_kernel<T, true>), dim3(grid),          \
                       dim3(256), 0, stream, (const T*)y, (T*)out, b, t, to,  \
                       kt, st, p0, o, hw);                                    \
  else                                                                        \
    hipLaunchKernelGGL((temporal_merge_kernel<T, false>), dim3(grid),         \
                       dim3(256), 0, stream, (const T*)y, (T*)out, b, t, to,  \
                       kt, st, p0, o, hw);
  switch (dtype) {
    case VFA_F32: VFA_TM_CASE(float) break;
    case VFA_BF16: VFA_TM_CASE(__hip_bfloat16) break;
    case VFA_F16: VFA_TM_CASE(__half) break;
  }
#undef VFA_TM_CASE
}

}  // extern "C"
