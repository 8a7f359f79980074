// Fused MFMA linear kernel for gfx950:  C = act(A @ W^T + bias), bf16.
//
//   A (M, K) row-major, W (N, K) row-major (torch nn.Linear layout), both
//   bf16; C (M, N) bf16; f32 accumulation; bias + activation applied in
//   the epilogue (QuickGELU / tanh-GELU / ReLU / none) — the separate
//   activation kernel's full HBM round trip disappears.
//
// Structure (cdna_hip_programming.md §5 ladder step 3 + st_16x32 swizzle):
//   128x128 tile, BK = 64, 4 waves (2M x 2N), 64x64 C per wave as a 4x4
//   grid of v_mfma_f32_16x16x32_bf16 accumulators.
//   Both operands stage through LDS as [row][64] bf16 images (A rows = m,
//   B rows = n; the B fragment of C[m][n] = dot_k A[m][k] W[n][k] reads
//   the same row-major image as A).  Staging uses
//   __builtin_amdgcn_global_load_lds width 16 (2 LDS buffers, next K-tile
//   in flight during compute), with the st_16x32 XOR swizzle
//   (byte ^= ((byte>>9)&1)<<5 inside each 1024 B subtile) applied to the
//   *global source* address so the LDS image stays lane-linear for glds;
//   ds_read_b128 fragment reads apply the same XOR.
//   Partial tiles (M tail) take a bounds-checked vector-staging path with
//   identical LDS image.
#include "vfa_common.h"

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

namespace {

constexpr int BM = 128, BN = 128, BK = 64;
constexpr int TILE_B = BM * BK * 2;            // bytes per operand tile
constexpr int NSUB = TILE_B / 1024;            // 16 glds subtiles / operand

__device__ __forceinline__ int swz(int byte_off) {
  return byte_off ^ (((byte_off >> 9) & 1) << 5);
}

__device__ __forceinline__ float act_f(float x, int kind) {
  if (kind == 1) return fmaxf(x, 0.f);
  if (kind == 2) return x / (1.0f + __expf(-1.702f * x));   // QuickGELU
  if (kind == 3) {
    const float k0 = 0.7978845608028654f, k1 = 0.044715f;
    return 0.5f * x * (1.0f + tanhf(k0 * (x + k1 * x * x * x)));
  }
  return x;
}

// stage a (rows x 64) bf16 tile into an LDS image with the st_16x32
// swizzle via glds: each wave covers NSUB/4 1024-B subtiles (8 rows each).
__device__ __forceinline__ void stage_glds(const __bf16* __restrict__ g,
                                           long long row_stride_elems,
                                           char* lds_base, int wave,
                                           int lane) {
  const int off = lane * 16;                   // linear LDS offset in subtile
  const int off_log = off ^ (((off >> 9) & 1) << 5);
  const int r_in = off_log >> 7;               // row within subtile
  const int b_in = off_log & 127;              // byte within 128-B row
#pragma unroll
  for (int i = 0; i < NSUB / 4; ++i) {
    const int sub = wave * (NSUB / 4) + i;
    const __bf16* src = g + (long long)(sub * 8 + r_in) * row_stride_elems;
    // LDS destination is wave-uniform base + lane*16 (hardware-added);
    // the per-lane *global* address carries the swizzle
    __builtin_amdgcn_global_load_lds(
        reinterpret_cast<const unsigned int*>(
            reinterpret_cast<const char*>(src) + b_in),
        reinterpret_cast<unsigned int*>(lds_base + sub * 1024), 16, 0, 0);
  }
}

// bounds-checked fallback staging (M tail): same LDS image, zero fill.
__device__ __forceinline__ void stage_guard(const __bf16* __restrict__ g,
                                            long long row_stride_elems,
                                            int valid_rows, char* lds_base,
                                            int tid) {
  for (int t = tid; t < BM * 8; t += 256) {    // 8 x 16-B segments per row
    const int row = t >> 3, seg = t & 7;
    uint4 v = {0u, 0u, 0u, 0u};
    if (row < valid_rows)
      v = *reinterpret_cast<const uint4*>(g + row * row_stride_elems +
                                          seg * 8);
    *reinterpret_cast<uint4*>(lds_base + swz(row * 128 + seg * 16)) = v;
  }
}

template <int ACT, bool FULL>
__global__ __launch_bounds__(256)
void linear_act_kernel(const __bf16* __restrict__ a,
                       const __bf16* __restrict__ w,
                       const __bf16* __restrict__ bias,
                       __bf16* __restrict__ c, int m, int n, int k,
                       int tiles_m, int tiles_n) {
  extern __shared__ __attribute__((aligned(1024))) char smem[];
  // layout: [buf][A | B], each TILE_B bytes
  auto sA = [&](int buf) { return smem + buf * 2 * TILE_B; };
  auto sB = [&](int buf) { return smem + buf * 2 * TILE_B + TILE_B; };

  // XCD-aware bijective remap (consecutive remapped ids share an XCD)
  const int nwg = tiles_m * tiles_n;
  int wg = blockIdx.x;
  {
    const int xcd = wg % 8, orig = wg / 8;
    const int q = nwg / 8, r = nwg % 8;
    wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + orig;
  }
  const int tile_n = wg / tiles_m, tile_m = wg % tiles_m;
  const int m0 = tile_m * BM, n0 = tile_n * BN;

  const int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  const int lo = lane & 15, hi4 = lane >> 4;
  const int wm = wave >> 1, wn = wave & 1;     // 2x2 wave grid

  const int nk = k / BK;
  const int valid_m = m - m0;

  if (FULL) {
    stage_glds(a + (long long)m0 * k, k, sA(0), wave, lane);
    stage_glds(w + (long long)n0 * k, k, sB(0), wave, lane);
  } else {
    stage_guard(a + (long long)m0 * k, k, valid_m, sA(0), threadIdx.x);
    stage_guard(w + (long long)n0 * k, k, n - n0, sB(0), threadIdx.x);
  }

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  for (int kt = 0; kt < nk; ++kt) {
    __syncthreads();                           // tile kt resident
    const int cur = kt & 1;
    if (kt + 1 < nk) {
      const int nxt = 1 - cur;
      if (FULL) {
        stage_glds(a + (long long)m0 * k + (kt + 1) * BK, k, sA(nxt), wave,
                   lane);
        stage_glds(w + (long long)n0 * k + (kt + 1) * BK, k, sB(nxt), wave,
                   lane);
      } else {
        stage_guard(a + (long long)m0 * k + (kt + 1) * BK, k, valid_m,
                    sA(nxt), threadIdx.x);
        stage_guard(w + (long long)n0 * k + (kt + 1) * BK, k, n - n0,
                    sB(nxt), threadIdx.x);
      }
    }
    // compute on tile kt
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {           // two 32-deep MFMA steps
      bf16x8 afr[4], bfr[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int arow = wm * 64 + i * 16 + lo;
        afr[i] = *reinterpret_cast<const bf16x8*>(
            sA(cur) + swz(arow * 128 + kk * 64 + hi4 * 16));
      }
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int brow = wn * 64 + j * 16 + lo;
        bfr[j] = *reinterpret_cast<const bf16x8*>(
            sB(cur) + swz(brow * 128 + kk * 64 + hi4 * 16));
      }
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afr[i], bfr[j], acc[i][j], 0, 0, 0);
    }
    // no trailing barrier: the next iteration's top __syncthreads() both
    // drains the in-flight glds and orders reads before buffer reuse
  }

  // epilogue: bias + activation, bf16 store
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    const int col = n0 + wn * 64 + j * 16 + lo;
    const float bv = bias ? (float)bias[col] : 0.f;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = m0 + wm * 64 + i * 16 + hi4 * 4 + r;
        if (!FULL && row >= m) continue;
        c[(long long)row * n + col] =
            (__bf16)act_f(acc[i][j][r] + bv, ACT);
      }
    }
  }
}

template <int ACT>
void launch_linear(const void* a, const void* w, const void* bias, void* c,
                   int m, int n, int k, hipStream_t stream) {
  const int tiles_m = (m + BM - 1) / BM, tiles_n = n / BN;
  const dim3 grid(tiles_m * tiles_n);
  const size_t lds = 4 * TILE_B;               // 2 buffers x (A + B)
  if (m % BM == 0)
    hipLaunchKernelGGL((linear_act_kernel<ACT, true>), grid, dim3(256), lds,
                       stream, (const __bf16*)a, (const __bf16*)w,
                       (const __bf16*)bias, (__bf16*)c, m, n, k, tiles_m,
                       tiles_n);
  else
    hipLaunchKernelGGL((linear_act_kernel<ACT, false>), grid, dim3(256), lds,
                       stream, (const __bf16*)a, (const __bf16*)w,
                       (const __bf16*)bias, (__bf16*)c, m, n, k, tiles_m,
                       tiles_n);
}

}  // namespace

extern "C" {

// act: 0 none, 1 relu, 2 quick_gelu, 3 gelu_tanh.  Requires K % 64 == 0,
// N % 128 == 0 (checked at the binding).
void vfa_linear_act(const void* a, const void* w, const void* bias, void* c,
                    int m, int n, int k, int act, hipStream_t stream) {
  switch (act) {
    case 0: launch_linear<0>(a, w, bias, c, m, n, k, stream); break;
    case 1: launch_linear<1>(a, w, bias, c, m, n, k, stream); break;
    case 2: launch_linear<2>(a, w, bias, c, m, n, k, stream); break;
    case 3: launch_linear<3>(a, w, bias, c, m, n, k, stream); break;
  }
}

}  // extern "C"
