// Fused elementwise activations for gfx950 — vectorized 16 B/lane loads
// (cdna_hip_programming.md G13: scalar bf16 loads cost ~2-2.5x).
#include "vfa_common.h"

namespace {

__device__ __forceinline__ float quick_gelu_f(float x) {
  // CLIP QuickGELU: x * sigmoid(1.702 x)
  return x / (1.0f + __expf(-1.702f * x));
}

__device__ __forceinline__ float gelu_tanh_f(float x) {
  const float k0 = 0.7978845608028654f;   // sqrt(2/pi)
  const float k1 = 0.044715f;
  float t = tanhf(k0 * (x + k1 * x * x * x));
  return 0.5f * x * (1.0f + t);
}

// VEC elements per thread; T is the storage type. Grid-stride over n/VEC.
template <typename T, int VEC, int KIND>  // KIND 0=quick_gelu 1=gelu_tanh
__global__ void act_kernel(const T* __restrict__ in, T* __restrict__ out,
                           long long n) {
  using VecT = __attribute__((ext_vector_type(VEC * sizeof(T) / 4))) unsigned;
  long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i * VEC < n; i += stride) {
    long long base = i * VEC;
    if (base + VEC <= n) {
      VecT v = *reinterpret_cast<const VecT*>(in + base);
      T tmp[VEC];
      *reinterpret_cast<VecT*>(tmp) = v;
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        float x = to_f32<T>(tmp[j]);
        tmp[j] = from_f32<T>(KIND == 0 ? quick_gelu_f(x) : gelu_tanh_f(x));
      }
      *reinterpret_cast<VecT*>(out + base) = *reinterpret_cast<VecT*>(tmp);
    } else {
      for (long long j = base; j < n; ++j) {
        float x = to_f32<T>(in[j]);
        out[j] = from_f32<T>(KIND == 0 ? quick_gelu_f(x) : gelu_tanh_f(x));
      }
    }
  }
}

template <typename T, int KIND>
void launch_act(const void* in, void* out, long long n, hipStream_t stream) {
  constexpr int VEC = 16 / sizeof(T);
  long long nvec = (n + VEC - 1) / VEC;
  int block = 256;
  // one sweep, one iteration per thread where possible: the grid-stride
  // loop's serialized load->store chain leaves HBM underused at this size
  int grid = (int)min((nvec + block - 1) / block, (long long)65536);
  hipLaunchKernelGGL((act_kernel<T, VEC, KIND>), dim3(grid), dim3(block), 0,
                     stream, (const T*)in, (T*)out, n);
}

}  // namespace

extern "C" {

void vfa_quick_gelu(const void* in, void* out, long long n, int dtype,
                    hipStream_t stream) {
  switch (dtype) {
    case VFA_F32: launch_act<float, 0>(in, out, n, stream); break;
    case VFA_BF16: launch_act<__hip_bfloat16, 0>(in, out, n, stream); break;
    case VFA_F16: launch_act<__half, 0>(in, out, n, stream); break;
  }
}

void vfa_gelu_tanh(const void* in, void* out, long long n, int dtype,
                   hipStream_t stream) {
  switch (dtype) {
    case VFA_F32: launch_act<float, 1>(in, out, n, stream); break;
    case VFA_BF16: launch_act<__hip_bfloat16, 1>(in, out, n, stream); break;
    case VFA_F16: launch_act<__half, 1>(in, out, n, stream); break;
  }
}

}  // extern "C"
