// Implicit-GEMM conv2d for gfx950 (CDNA4), NHWC bf16 — hand-written MFMA
// replacement for MIOpen's igemm on the 3x3 / 1x5 / 5x1 / 7x7 hot shapes
// of the ResNet / RAFT / I3D(flattened) / VGGish conv stacks (reference
// conv stacks: models/resnet & models/raft/raft_src/extractor.py:118-192,
// models/i3d/i3d_src/i3d_net.py:37-105, vggish_src/vggish.py:108-118).
//
// GEMM view:  C[M, N] = act( A[M, Kr] @ W[N, Kr]^T + bias [+ res] )
//   M  = B*OH*OW output pixels, N = K_out channels,
//   Kr = KH*KW*C with reduction index k = (r*KW + s)*C + c.
// A is the im2col view of the PRE-PADDED input (B, Hp, Wp, C) — never
// materialized: each staged A row is an input pixel's C-slice at tap
// (r, s), address  base(m) + (r*Wp + s)*C + c  with
// base(m) = ((b*Hp + oy*sh)*Wp + ox*sw)*C precomputed PER THREAD before
// the K-loop (each thread always stages the same tile rows; only the
// k-offset changes per K-tile).  W is torch's channels_last conv weight —
// physically (K_out, KH, KW, C) = exactly the (N, Kr) row-major B operand,
// so weights need NO reshuffling and B staging is identical to the linear
// kernel's.
//
// Machinery shared with gemm.hip (cdna_hip_programming.md §5): 256x256 /
// 128x128 tiles, BK=64, double-buffered LDS staged by
// __builtin_amdgcn_global_load_lds width 16 with the conflict-free XOR
// swizzle (cswz) on the per-lane *source* address (lane-linear LDS
// image), glds for
// the next K-tile spread across the current tile's two MFMA half-steps,
// mfma_f32_16x16x32_bf16, fused bias+activation(+residual) epilogue.
// Requires C % 8 == 0 (each lane's 16-B segment stays inside one tap).
#include "vfa_common.h"

typedef __bf16 bf16x8c __attribute__((ext_vector_type(8)));
typedef float f32x4c __attribute__((ext_vector_type(4)));

namespace {

constexpr int BK = 64;

// see gemm.hip swz(): conflict-free row-slot XOR ((row>>1)&7, searched
// against gfx950's REAL mixed b128 lane groups) for the 128-B-row images
__device__ __forceinline__ int cswz(int byte_off) {
  return byte_off ^ (((byte_off >> 8) & 7) << 4);
}

__device__ __forceinline__ float conv_act_f(float x, int kind) {
  if (kind == 1) return fmaxf(x, 0.f);
  if (kind == 2) return x / (1.0f + __expf(-1.702f * x));  // QuickGELU
  if (kind == 3) {
    const float k0 = 0.7978845608028654f, k1 = 0.044715f;
    return 0.5f * x * (1.0f + tanhf(k0 * (x + k1 * x * x * x)));
  }
  if (kind == 4) return fmaxf(x, 0.f) + 0.1f * fminf(x, 0.f);  // LeakyReLU .1
  return x;
}

struct ConvGeom {
  int b, hp, wp, cin;       // input dims as addressed (REAL h/w when the
                            // kernel pads inline; pre-padded dims when the
                            // pad was materialized, with pt = pl = 0)
  int oh, ow, kout;         // output
  int kh, kw, sh, sw;       // filter / stride
  int pt, pl;               // inline zero-pad (top/left); bottom/right are
                            // implied by oh/ow + the validity check
  int kr;                   // KH*KW*C
};

// decompose output-pixel index m -> (image base, receptive-field origin)
struct PixRef {
  long long base;           // bi * H * W * C
  int y0, x0;               // oy*sh - pt, ox*sw - pl (may be negative)
};

__device__ __forceinline__ PixRef pix_ref(const ConvGeom& g, int m) {
  const int ox = m % g.ow;
  const int t = m / g.ow;
  const int oy = t % g.oh;
  const int bi = t / g.oh;
  return PixRef{(long long)bi * g.hp * g.wp * g.cin,
                oy * g.sh - g.pt, ox * g.sw - g.pl};
}

// Stage the (ROWS x 64) A-tile via glds: per-thread row refs precomputed;
// the per-lane k position picks tap (r, s) and channel; out-of-image taps
// redirect the load to a zero page (the inline zero-padding — no
// materialized pad pass).  piece splits the wave's glds issues across MFMA
// half-steps (-1 = all).
template <int ROWS, int WAVES>
__device__ __forceinline__ void conv_stage_a(
    const __bf16* __restrict__ x, const __bf16* __restrict__ zp,
    const ConvGeom& g, const PixRef* __restrict__ aref, int k0,
    char* lds_base, int wave, int lane, int kfrac0, int arin_half,
    int piece, int npieces) {
  constexpr int NSUB = ROWS * 128 / 1024;
  constexpr int PER_WAVE = NSUB / WAVES;
#pragma unroll
  for (int i = 0; i < PER_WAVE; ++i) {
    if (piece >= 0 && (i * npieces) / PER_WAVE != piece) continue;
    const int sub = wave * PER_WAVE + i;
    // per-subtile swizzle: f = ((sub&1)<<2)|(r_in>>1) on element bits 3..5
    const int k = k0 + (kfrac0 ^
                        ((((sub & 1) << 2) | arin_half) << 3));
    const int tap = k / g.cin;          // uniform-cost u32 div
    const int c = k - tap * g.cin;
    const int r = tap / g.kw;
    const int s = tap - r * g.kw;
    const int iy = aref[i].y0 + r, ix = aref[i].x0 + s;
    const bool ok = (unsigned)iy < (unsigned)g.hp &&
                    (unsigned)ix < (unsigned)g.wp;
    const __bf16* src =
        ok ? x + aref[i].base + ((long long)iy * g.wp + ix) * g.cin + c
           : zp + (lane & 63) * 8;
    __builtin_amdgcn_global_load_lds(
        reinterpret_cast<const unsigned int*>(src),
        reinterpret_cast<unsigned int*>(lds_base + sub * 1024), 16, 0, 0);
  }
}

// Weight staging == linear kernel's stage_glds (W is (N, Kr) row-major).
template <int ROWS, int WAVES>
__device__ __forceinline__ void conv_stage_w(
    const __bf16* __restrict__ wgt, long long row_stride, char* lds_base,
    int wave, int lane, int piece, int npieces) {
  constexpr int NSUB = ROWS * 128 / 1024;
  constexpr int PER_WAVE = NSUB / WAVES;
  // subtile-local swizzle; f = ((sub&1)<<2)|(r_in>>1) (see gemm.hip)
  const int off = lane * 16;
  const int r_in = off >> 7;
#pragma unroll
  for (int i = 0; i < PER_WAVE; ++i) {
    if (piece >= 0 && (i * npieces) / PER_WAVE != piece) continue;
    const int sub = wave * PER_WAVE + i;
    const int b_in = (off & 127) ^
                     ((((sub & 1) << 2) | (r_in >> 1)) << 4);
    const __bf16* src = wgt + (long long)(sub * 8 + r_in) * row_stride;
    __builtin_amdgcn_global_load_lds(
        reinterpret_cast<const unsigned int*>(
            reinterpret_cast<const char*>(src) + b_in),
        reinterpret_cast<unsigned int*>(lds_base + sub * 1024), 16, 0, 0);
  }
}

// Bounds-checked A staging (M-edge tiles and the ragged last K-tile):
// same LDS image, zero fill (covers out-of-image taps too).
template <int ROWS, int THREADS>
__device__ __forceinline__ void conv_guard_a(
    const __bf16* __restrict__ x, const ConvGeom& g, int m0, int mtotal,
    int k0, char* lds_base, int tid) {
  for (int t = tid; t < ROWS * 8; t += THREADS) {  // 8 16-B segs per row
    const int row = t >> 3, seg = t & 7;
    uint4 v = {0u, 0u, 0u, 0u};
    const int m = m0 + row;
    const int k = k0 + seg * 8;
    if (m < mtotal && k < g.kr) {
      const PixRef pr = pix_ref(g, m);
      const int tap = k / g.cin;
      const int c = k - tap * g.cin;
      const int r = tap / g.kw;
      const int s = tap - r * g.kw;
      const int iy = pr.y0 + r, ix = pr.x0 + s;
      if ((unsigned)iy < (unsigned)g.hp && (unsigned)ix < (unsigned)g.wp)
        v = *reinterpret_cast<const uint4*>(
            x + pr.base + ((long long)iy * g.wp + ix) * g.cin + c);
    }
    *reinterpret_cast<uint4*>(lds_base + cswz(row * 128 + seg * 16)) = v;
  }
}

template <int ROWS, int THREADS>
__device__ __forceinline__ void conv_guard_w(
    const __bf16* __restrict__ wgt, long long row_stride, int n0, int ntotal,
    int k0, int kr, char* lds_base, int tid) {
  for (int t = tid; t < ROWS * 8; t += THREADS) {
    const int row = t >> 3, seg = t & 7;
    uint4 v = {0u, 0u, 0u, 0u};
    if (n0 + row < ntotal && k0 + seg * 8 < kr) {
      v = *reinterpret_cast<const uint4*>(wgt + (long long)row * row_stride +
                                          seg * 8);
    }
    *reinterpret_cast<uint4*>(lds_base + cswz(row * 128 + seg * 16)) = v;
  }
}

// Tile variants: 256x256 (8 waves, 2Mx4N), 128x128 (4 waves, 2x2),
// 256x64 (4 waves, 4Mx1N — for K_out = 64/192 nets where a 128-wide N
// tile is half empty).
template <int ACT, int BM, int BN, int WAVES, int WN>
__global__ __launch_bounds__(WAVES * 64)
void conv2d_nhwc_kernel(const __bf16* __restrict__ x,
                        const __bf16* __restrict__ wgt,
                        const __bf16* __restrict__ bias,
                        const __bf16* __restrict__ res,
                        const __bf16* __restrict__ zp,
                        __bf16* __restrict__ out, ConvGeom g, int mtotal,
                        int ldc, int tiles_m, int tiles_n) {
  constexpr int MI = BM / (WAVES / WN) / 16;   // 16-row MFMA tiles / wave
  constexpr int NJ = BN / WN / 16;             // 16-col MFMA tiles / wave
  static_assert(NJ == 4, "epilogue assumes 4 column fragments per wave");
  constexpr int TILE_A = BM * BK * 2, TILE_B = BN * BK * 2;
  constexpr int THREADS = WAVES * 64;
  constexpr int PER_WAVE = (BM * 128 / 1024) / WAVES;

  extern __shared__ __attribute__((aligned(1024))) char smem[];
  auto sA = [&](int buf) { return smem + buf * (TILE_A + TILE_B); };
  auto sB = [&](int buf) { return smem + buf * (TILE_A + TILE_B) + TILE_A; };

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
  const int wm = wave / WN, wn = wave % WN;

  const int kr = g.kr, n = g.kout;
  const int nk = (kr + BK - 1) / BK;
  const int valid_m = mtotal - m0, valid_n = n - n0;
  const bool tile_full = valid_m >= BM && valid_n >= BN;
  const bool kfull = (kr % BK) == 0;

  // per-thread A row refs for the glds path (same rows every K-tile);
  // the per-lane swizzled byte offset contributes kfrac elements
  const int a_rin = (lane * 16) >> 7;
  const int kfrac0 = ((lane * 16) & 127) >> 1;   // pre-swizzle elements
  PixRef aref[PER_WAVE];
#pragma unroll
  for (int i = 0; i < PER_WAVE; ++i) {
    const int sub = wave * PER_WAVE + i;
    const int m = m0 + sub * 8 + a_rin;
    aref[i] = pix_ref(g, m < mtotal ? m : mtotal - 1);
  }

  auto stage = [&](int buf, int kt) {
    const int k0 = kt * BK;
    if (tile_full && (kfull || kt + 1 < nk)) {
      conv_stage_a<BM, WAVES>(x, zp, g, aref, k0, sA(buf), wave, lane,
                              kfrac0, a_rin >> 1, -1, 1);
      conv_stage_w<BN, WAVES>(wgt + (long long)n0 * kr + k0, kr, sB(buf),
                              wave, lane, -1, 1);
    } else {
      conv_guard_a<BM, THREADS>(x, g, m0, mtotal, k0, sA(buf), threadIdx.x);
      conv_guard_w<BN, THREADS>(wgt + (long long)n0 * kr + k0, kr, n0, n,
                                k0, kr, sB(buf), threadIdx.x);
    }
  };

  stage(0, 0);

  f32x4c acc[MI][NJ];
#pragma unroll
  for (int i = 0; i < MI; ++i)
#pragma unroll
    for (int j = 0; j < NJ; ++j) acc[i][j] = f32x4c{0.f, 0.f, 0.f, 0.f};

  for (int kt = 0; kt < nk; ++kt) {
    __syncthreads();
    const int cur = kt & 1;
    const int k0_nxt = (kt + 1) * BK;
    const bool glds_nxt = (kt + 1 < nk) && tile_full &&
                          (kfull || kt + 2 < nk);
    if ((kt + 1 < nk) && !glds_nxt) stage(1 - cur, kt + 1);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8c afr[MI], bfr[NJ];
#pragma unroll
      for (int j = 0; j < NJ; ++j) {
        const int brow = wn * (NJ * 16) + j * 16 + lo;
        bfr[j] = *reinterpret_cast<const bf16x8c*>(
            sB(cur) + cswz(brow * 128 + kk * 64 + hi4 * 16));
      }
#pragma unroll
      for (int i = 0; i < MI; ++i) {
        const int arow = wm * (MI * 16) + i * 16 + lo;
        afr[i] = *reinterpret_cast<const bf16x8c*>(
            sA(cur) + cswz(arow * 128 + kk * 64 + hi4 * 16));
      }
      if (glds_nxt) {
        conv_stage_a<BM, WAVES>(x, zp, g, aref, k0_nxt, sA(1 - cur), wave,
                                lane, kfrac0, a_rin >> 1, kk, 2);
        conv_stage_w<BN, WAVES>(wgt + (long long)n0 * kr + k0_nxt, kr,
                                sB(1 - cur), wave, lane, kk, 2);
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int i = 0; i < MI; ++i)
#pragma unroll
        for (int j = 0; j < NJ; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afr[i], bfr[j], acc[i][j], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
  }

#pragma unroll
  for (int j = 0; j < NJ; ++j) {
    const int col = n0 + wn * (NJ * 16) + j * 16 + lo;
    if (!tile_full && col >= n) continue;
    const float bv = bias ? (float)bias[col] : 0.f;
#pragma unroll
    for (int i = 0; i < MI; ++i) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = m0 + wm * (MI * 16) + i * 16 + hi4 * 4 + r;
        if (!tile_full && row >= mtotal) continue;
        float v = acc[i][j][r] + bv;
        if (res) v += (float)res[(long long)row * n + col];
        // ldc > n: writing a channel slice of a wider NHWC buffer (the
        // caller eliminated a cat copy); res stays logically (M, n)
        out[(long long)row * ldc + col] = (__bf16)conv_act_f(v, ACT);
      }
    }
  }
}

template <int ACT, int BM, int BN, int WAVES, int WN>
void launch_cfg(const void* x, const void* w, const void* bias,
                const void* res, const void* zp, void* out,
                const ConvGeom& g, int m, int ldc, hipStream_t stream) {
  const int tiles_m = (m + BM - 1) / BM, tiles_n = (g.kout + BN - 1) / BN;
  const dim3 grid(tiles_m * tiles_n);
  const size_t lds = 2 * (size_t)(BM + BN) * BK * 2;
  hipLaunchKernelGGL((conv2d_nhwc_kernel<ACT, BM, BN, WAVES, WN>), grid,
                     dim3(WAVES * 64), lds, stream, (const __bf16*)x,
                     (const __bf16*)w, (const __bf16*)bias,
                     (const __bf16*)res, (const __bf16*)zp, (__bf16*)out,
                     g, m, ldc, tiles_m, tiles_n);
}

template <int ACT>
void launch_conv(const void* x, const void* w, const void* bias,
                 const void* res, const void* zp, void* out,
                 const ConvGeom& g, int ldc, hipStream_t stream) {
  const int m = g.b * g.oh * g.ow;
  // pick the tile minimizing padded work, preferring the bigger tile when
  // waste ties and the grid still fills the chip (256 CUs)
  auto cost = [&](int bm, int bn, long long min_wgs) {
    const long long tm = (m + bm - 1) / bm, tn = (g.kout + bn - 1) / bn;
    long long padded = tm * bm * tn * bn;
    if (tm * tn < min_wgs) padded = padded * 4;   // fill penalty
    return padded;
  };
  const long long c256 = (g.kout >= 256 && m >= 4096)
                             ? cost(256, 256, 150) : (1LL << 62);
  const long long c128 = cost(128, 128, 120);
  const long long c64 = cost(256, 64, 120);
  if (c256 <= c128 && c256 <= c64)
    launch_cfg<ACT, 256, 256, 8, 4>(x, w, bias, res, zp, out, g, m, ldc, stream);
  else if (c64 < c128)
    launch_cfg<ACT, 256, 64, 4, 1>(x, w, bias, res, zp, out, g, m, ldc, stream);
  else
    launch_cfg<ACT, 128, 128, 4, 2>(x, w, bias, res, zp, out, g, m, ldc, stream);
}

// ------------------------------------------------------------- pad kernel
// zero-pad H/W and optionally CHANNELS of an NHWC tensor:
// (B, H, W, Cin) -> (B, H+pt+pb, W+pl+pr, Cout), Cout % 8 == 0, zeros for
// c >= Cin.  The channel pad is how C=3/C=2 stems (ResNet/RAFT/I3D) and
// the C=324 RAFT corr input reach the implicit-GEMM kernel (which needs
// C % 8 == 0).  Memory-bound; 16-B vectorized, elementwise on the one
// segment straddling Cin.
__global__ void pad2d_nhwc_kernel(const __bf16* __restrict__ x,
                                  __bf16* __restrict__ out, int b, int h,
                                  int w, int cin, int cout, int pt, int pb,
                                  int pl, int pr) {
  const int hp = h + pt + pb, wp = w + pl + pr;
  const long long total = (long long)b * hp * wp * (cout / 8);
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < total; i += (long long)gridDim.x * blockDim.x) {
    const int cseg = (int)(i % (cout / 8));
    long long t = i / (cout / 8);
    const int xw = (int)(t % wp);
    t /= wp;
    const int xh = (int)(t % hp);
    const int bi = (int)(t / hp);
    uint4 v = {0u, 0u, 0u, 0u};
    const int sy = xh - pt, sx = xw - pl;
    if (sy >= 0 && sy < h && sx >= 0 && sx < w) {
      const long long src = (((long long)bi * h + sy) * w + sx) * cin;
      if (cseg * 8 + 8 <= cin) {
        v = *reinterpret_cast<const uint4*>(x + src + cseg * 8);
      } else if (cseg * 8 < cin) {
        __bf16* e = reinterpret_cast<__bf16*>(&v);
        for (int j = 0; cseg * 8 + j < cin; ++j)
          e[j] = x[src + cseg * 8 + j];
      }
    }
    *reinterpret_cast<uint4*>(
        out + (((long long)bi * hp + xh) * wp + xw) * cout + cseg * 8) = v;
  }
}

}  // namespace

extern "C" {

// act: 0 none, 1 relu, 2 quick_gelu, 3 gelu_tanh, 4 leaky_relu(0.1)
// pt/pl: inline zero-pad applied by the kernel (zp = 1 KiB zero page);
// pass 0 with pre-padded h/w when the pad was materialized
// ldc: output row stride in elements (= kout normally; larger when the
// epilogue writes a channel slice of a wider NHWC buffer)
void vfa_conv2d_nhwc(const void* x, const void* w, const void* bias,
                     const void* res, const void* zp, void* out, int b,
                     int hp, int wp, int cin, int oh, int ow, int kout,
                     int kh, int kw, int sh, int sw, int pt, int pl,
                     int ldc, int act, hipStream_t stream) {
  ConvGeom g{b, hp, wp, cin, oh, ow, kout, kh, kw, sh, sw, pt, pl,
             kh * kw * cin};
  switch (act) {
    case 0: launch_conv<0>(x, w, bias, res, zp, out, g, ldc, stream); break;
    case 1: launch_conv<1>(x, w, bias, res, zp, out, g, ldc, stream); break;
    case 2: launch_conv<2>(x, w, bias, res, zp, out, g, ldc, stream); break;
    case 3: launch_conv<3>(x, w, bias, res, zp, out, g, ldc, stream); break;
    case 4: launch_conv<4>(x, w, bias, res, zp, out, g, ldc, stream); break;
  }
}

void vfa_pad2d_nhwc(const void* x, void* out, int b, int h, int w, int cin,
                    int cout, int pt, int pb, int pl, int pr,
                    hipStream_t stream) {
  const long long total = (long long)b * (h + pt + pb) * (w + pl + pr) *
                          (cout / 8);
  const int threads = 256;
  const int blocks = (int)min((total + threads - 1) / threads, 2048LL);
  hipLaunchKernelGGL(pad2d_nhwc_kernel, dim3(blocks), dim3(threads), 0,
                     stream, (const __bf16*)x, (__bf16*)out, b, h, w, cin,
                     cout, pt, pb, pl, pr);
}

}  // extern "C"
