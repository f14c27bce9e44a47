// Hand-written MFMA implicit-GEMM 3x3 convolution — gfx950 (CDNA4).
//
// NHWC bf16, pad=1, stride 1 or 2, fp32 MFMA accumulation, bf16 out.
// GEMM view (never materialized):  y[M, Co] = A[M, 9*Ci] @ W[Co, 9*Ci]^T
// with M = N*Ho*Wo and A the im2col of x.  Requires Ci % 64 == 0 so a
// BK=64 k-step stays inside one (r, s) filter tap — then every A-tile
// row is a CONTIGUOUS 64-channel slice of x (vector loads, one
// whole-row pad/bounds guard, no per-element guards — the guide's
// trap 4c).  Weights are laid out [Co][3][3][Ci] (the channels_last
// view of a PyTorch Conv2d weight), so B rows are contiguous too.
//
// Structure: 128x128 output tile, 256 threads = 4 waves in 2x2 (each
// wave 64x64 = 4x4 fragments of v_mfma_f32_16x16x32_bf16), register-
// staged double-buffered LDS with +8 bf16 row padding (the validated
// v2 GEMM geometry, gemm1x1_kernels.hip), XCD-aware block->tile remap.
// The m -> (n, ho, wo) decode happens ONCE per tile per thread (the
// 4 staged rows are fixed), so the k-loop does only adds and guards.
//
// dgrad (stride 1) reuses THIS kernel: dx = conv3x3(dy, w_rot) with
// w_rot[ci][r'][s'][co] = w[co][2-r'][2-s'][ci] (rotated + transposed,
// materialized by the Python layer — it is only the small weight).
//
// Replaces MIOpen's igemm kernels on the ResNet-50 3x3 hot path
// (reference workload: gossip_sgd.py:693-707 via torchvision/cuDNN).

#include <hip/hip_runtime.h>

#include <cstdint>

namespace {

typedef unsigned short ushort_t;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

__device__ __forceinline__ ushort_t f2b_(float f) {
  union { unsigned int i; float f; } v;
  v.f = f;
  unsigned int r = v.i + 0x7FFFu + ((v.i >> 16) & 1u);
  return (ushort_t)(r >> 16);
}

#define CBM 128
#define CBN 128
#define CBK 64
#define CSTRIDE 72  // 64 + 8 bf16 pad per LDS row

__device__ __forceinline__ int64_t conv_xcd_bid() {
  const int g8 = gridDim.x >> 3;
  return (int64_t)(blockIdx.x & 7) * g8 + (blockIdx.x >> 3);
}

// SPLITK=true: the 9*Ci reduction is partitioned over `split` block
// groups writing fp32 partials [split][M][Co] (reduced to bf16 by
// k_conv3x3_reduce) — recovers chip fill on small-M shapes (e.g. the
// 7x7 ResNet stage has only 52 output tiles for 512+ block slots).
template <int STRIDE, bool SPLITK>
__global__ __launch_bounds__(256) void k_conv3x3_nhwc_bf16(
    const ushort_t* __restrict__ X,  // [N, H, W, Ci]
    const ushort_t* __restrict__ Wt, // [Co, 3, 3, Ci]
    ushort_t* __restrict__ Y,        // [N, Ho, Wo, Co]
    float* __restrict__ P,           // [split, M, Co] (SPLITK only)
    int Nb, int H, int W, int Ci, int Co, int Ho, int Wo, int split) {
  __shared__ ushort_t As[2][CBM * CSTRIDE];
  __shared__ ushort_t Bs[2][CBN * CSTRIDE];

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int wm = (wave >> 1) * 64;
  const int wn = (wave & 1) * 64;
  const int frow = lane & 15;
  const int fk0 = (lane >> 4) * 8;

  const int64_t M = (int64_t)Nb * Ho * Wo;
  const int K9 = 9 * Ci;
  const int KT_all = K9 / CBK;

  const int n_tiles = (Co + CBN - 1) / CBN;
  const int64_t m_tiles = (M + CBM - 1) / CBM;
  const int64_t total_tiles = m_tiles * n_tiles * (SPLITK ? split : 1);

  // staging assignment: 1024 segments of 8 bf16 per operand tile;
  // thread t stages segs {t, t+256, t+512, t+768}: rows r8 = (t>>3) +
  // {0,32,64,96}, column c8 = (t&7)*8
  const int row0 = tid >> 3;
  const int c8 = (tid & 7) * 8;

  const int64_t bid0 = conv_xcd_bid();
  for (int64_t tile = bid0; tile < total_tiles; tile += gridDim.x) {
    int ksplit = 0;
    int64_t ct = tile;
    if (SPLITK) {
      ksplit = (int)(tile % split);
      ct = tile / split;
    }
    const int64_t tm = (ct / n_tiles) * CBM;
    const int tco = (int)(ct % n_tiles) * CBN;
    // k-step range of this split (CBK-aligned partition of KT_all)
    const int kt_lo = SPLITK ? (int)(((int64_t)KT_all * ksplit) / split) : 0;
    const int kt_hi =
        SPLITK ? (int)(((int64_t)KT_all * (ksplit + 1)) / split) : KT_all;
    const int KT = kt_hi - kt_lo;
    if (KT <= 0) continue;

    // per-tile row decode (hoisted out of the k-loop)
    int a_n[4], a_hi0[4], a_wi0[4];
    bool a_ok[4];
    const ushort_t* b_ptr[4];
    bool b_ok[4];
#pragma unroll
    for (int s = 0; s < 4; ++s) {
      const int row = row0 + s * 32;
      const int64_t m = tm + row;
      a_ok[s] = m < M;
      if (a_ok[s]) {
        const int64_t hw = (int64_t)Ho * Wo;
        const int n = (int)(m / hw);
        const int rem = (int)(m - (int64_t)n * hw);
        const int ho = rem / Wo;
        const int wo = rem - ho * Wo;
        a_n[s] = n;
        a_hi0[s] = ho * STRIDE - 1;
        a_wi0[s] = wo * STRIDE - 1;
      } else {
        a_n[s] = 0; a_hi0[s] = -2; a_wi0[s] = -2;
      }
      const int co = tco + row;
      b_ok[s] = co < Co;
      b_ptr[s] = Wt + (int64_t)(b_ok[s] ? co : 0) * K9;
    }

    f32x4 acc[4][4];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j) acc[i][j] = (f32x4){0.f, 0.f, 0.f, 0.f};

    bf16x8 ra[4], rb[4];

  // branch-free guarded loads: the address is clamped to a safe base
  // when out of bounds and the value masked after -- loads always issue
  // (no divergent branch around them), so all 8 stay in flight and
  // hide behind the MFMA block
#define CONV_LOAD(kt)                                                      \
  do {                                                                     \
    const int kk = (kt)*CBK;                                               \
    const int rs = kk / Ci;                                                \
    const int ci0 = kk - rs * Ci;                                          \
    const int r = rs / 3, sfs = rs - r * 3;                                \
    _Pragma("unroll") for (int s = 0; s < 4; ++s) {                        \
      const int hi = a_hi0[s] + r;                                         \
      const int wi = a_wi0[s] + sfs;                                       \
      const bool ok = a_ok[s] && (unsigned)hi < (unsigned)H                \
                      && (unsigned)wi < (unsigned)W;                       \
      const ushort_t* pa =                                                 \
          ok ? X + ((((int64_t)a_n[s] * H + hi) * W + wi) * Ci + ci0 + c8) \
             : X;                                                          \
      const bf16x8 va = *reinterpret_cast<const bf16x8*>(pa);              \
      const bf16x8 vb =                                                    \
          *reinterpret_cast<const bf16x8*>(b_ptr[s] + kk + c8);            \
      const bf16x8 z = (bf16x8){0, 0, 0, 0, 0, 0, 0, 0};                   \
      ra[s] = ok ? va : z;                                                 \
      rb[s] = b_ok[s] ? vb : z;                                            \
    }                                                                      \
  } while (0)

#define CONV_WRITE(buf)                                                    \
  do {                                                                     \
    _Pragma("unroll") for (int s = 0; s < 4; ++s) {                        \
      const int row = row0 + s * 32;                                       \
      *reinterpret_cast<bf16x8*>(As[buf] + row * CSTRIDE + c8) = ra[s];    \
      *reinterpret_cast<bf16x8*>(Bs[buf] + row * CSTRIDE + c8) = rb[s];    \
    }                                                                      \
  } while (0)

    CONV_LOAD(kt_lo);
    CONV_WRITE(0);

    for (int kt = 0; kt < KT; ++kt) {
      __syncthreads();
      const int buf = kt & 1;
      if (kt + 1 < KT) CONV_LOAD(kt_lo + kt + 1);  // issue early

#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        bf16x8 afrag[4], bfrag[4];
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          afrag[i] = *reinterpret_cast<const bf16x8*>(
              As[buf] + (wm + i * 16 + frow) * CSTRIDE + ks * 32 + fk0);
          bfrag[i] = *reinterpret_cast<const bf16x8*>(
              Bs[buf] + (wn + i * 16 + frow) * CSTRIDE + ks * 32 + fk0);
        }
#pragma unroll
        for (int i = 0; i < 4; ++i)
#pragma unroll
          for (int j = 0; j < 4; ++j)
            acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                afrag[i], bfrag[j], acc[i][j], 0, 0, 0);
      }

      __syncthreads();
      if (kt + 1 < KT) CONV_WRITE(buf ^ 1);
    }
#undef CONV_LOAD
#undef CONV_WRITE

#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j)
#pragma unroll
        for (int rr = 0; rr < 4; ++rr) {
          const int64_t gm = tm + wm + i * 16 + (lane >> 4) * 4 + rr;
          const int gc = tco + wn + j * 16 + (lane & 15);
          if (gm < M && gc < Co) {
            if (SPLITK)
              P[((int64_t)ksplit * M + gm) * Co + gc] = acc[i][j][rr];
            else
              Y[gm * Co + gc] = f2b_(acc[i][j][rr]);
          }
        }
  }
}

__global__ void k_conv3x3_reduce(const float* __restrict__ P,
                                 ushort_t* __restrict__ Y, int64_t numel,
                                 int split) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < numel; i += stride) {
    float acc = 0.f;
    for (int k = 0; k < split; ++k) acc += P[(int64_t)k * numel + i];
    Y[i] = f2b_(acc);
  }
}

inline int conv_grid(int64_t tiles) {
  if (tiles > 16384) tiles = 16384;
  int g = (int)((tiles + 7) & ~7);  // multiple of 8 for the XCD remap
  return g < 8 ? 8 : g;
}

// --------------------------------------------------- 3x3 wgrad (TN)
// dW[co, (r,s,ci)] = sum_m dy[m, co] * im2col(x)[m, (r,s,ci)] — the
// split-M TN GEMM (gemm1x1_kernels.hip wgrad geometry: m-fastest lane
// staging, transposed LDS images) with IMPLICIT im2col addressing on
// the B operand: per m-row decode (n,ho,wo), per column-group derive
// the filter tap (r,s) and channel ci0, pad-guard, one bf16x8 load.
// Ci % 64 == 0 keeps every 8-column group inside one tap.  fp32
// partials [split][Co][9Ci] + deterministic reduce.
#define WLDS_STRIDE 40  // 32 m-cols + 8 pad

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8w;
typedef __attribute__((ext_vector_type(4))) float f32x4w;

template <int STRIDE>
__global__ __launch_bounds__(256) void k_conv3x3_wgrad(
    const ushort_t* __restrict__ dy,  // [N, Ho, Wo, Co] (flat [M, Co])
    const ushort_t* __restrict__ X,   // [N, H, W, Ci]
    float* __restrict__ partials,     // [split][Co][9*Ci]
    int Nb, int H, int W, int Ci, int Co, int Ho, int Wo, int split) {
  __shared__ ushort_t As[2][128 * WLDS_STRIDE];  // [co][m] (x2 bufs)
  __shared__ ushort_t Bs[2][128 * WLDS_STRIDE];  // [col][m]

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int wm = (wave >> 1) * 64;  // co offset
  const int wn = (wave & 1) * 64;   // col offset
  const int frow = lane & 15;
  const int fk0 = (lane >> 4) * 8;

  const int64_t M = (int64_t)Nb * Ho * Wo;
  const int K9 = 9 * Ci;
  const int co_tiles = (Co + 127) / 128;
  const int col_tiles = (K9 + 127) / 128;
  const int64_t tiles = (int64_t)co_tiles * col_tiles * split;

  const int mrow = tid & 31;  // m-fastest in 16-lane groups
  const int cgrp = tid >> 5;  // 8 col-groups; this thread also +8

  const int64_t bid0 = conv_xcd_bid();
  for (int64_t t = bid0; t < tiles; t += gridDim.x) {
    const int s = (int)(t % split);
    const int64_t ct = t / split;
    const int tco = (int)(ct / col_tiles) * 128;
    const int tcol = (int)(ct % col_tiles) * 128;

    const int64_t m0 = (M * s) / split;
    const int64_t m1 = (M * (s + 1)) / split;

    // per-tile, per-half column decode (tap/ci0 fixed across k-steps)
    int h_r[2], h_s[2], h_ci0[2];
    bool h_ok[2];
#pragma unroll
    for (int half = 0; half < 2; ++half) {
      const int col = tcol + (cgrp + half * 8) * 8;
      h_ok[half] = col < K9;
      const int tap = h_ok[half] ? col / Ci : 0;
      h_r[half] = tap / 3;
      h_s[half] = tap - h_r[half] * 3;
      h_ci0[half] = h_ok[half] ? col - tap * Ci : 0;
    }
    const bool a_full = tco + 128 <= Co;

    f32x4w acc[4][4];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j)
        acc[i][j] = (f32x4w){0.f, 0.f, 0.f, 0.f};

    // register-staged double buffering (same as the 1x1 TN wgrad):
    // next chunk's loads issue right after the barrier
    bf16x8w ra[2], rb[2];

#define CW_LOAD(k0_)                                                       \
  do {                                                                     \
    const int64_t gm = (k0_) + mrow;                                       \
    const bool mok = gm < m1;                                              \
    int xn = 0, hi0 = -2, wi0 = -2;                                        \
    if (mok) {                                                             \
      const int64_t hw = (int64_t)Ho * Wo;                                 \
      xn = (int)(gm / hw);                                                 \
      const int rem = (int)(gm - (int64_t)xn * hw);                        \
      const int ho = rem / Wo;                                             \
      const int wo = rem - ho * Wo;                                        \
      hi0 = ho * STRIDE - 1;                                               \
      wi0 = wo * STRIDE - 1;                                               \
    }                                                                      \
    _Pragma("unroll") for (int half = 0; half < 2; ++half) {               \
      const int c8 = (cgrp + half * 8) * 8;                                \
      {                                                                    \
        const int gc0 = tco + c8;                                          \
        if (mok && a_full) {                                               \
          ra[half] = *reinterpret_cast<const bf16x8w*>(dy + gm * Co + gc0);\
        } else {                                                           \
          ushort_t tmp[8];                                                 \
          _Pragma("unroll") for (int j = 0; j < 8; ++j)                    \
            tmp[j] = (mok && gc0 + j < Co)                                 \
                ? dy[gm * Co + gc0 + j] : (ushort_t)0;                     \
          ra[half] = *reinterpret_cast<bf16x8w*>(tmp);                     \
        }                                                                  \
      }                                                                    \
      {                                                                    \
        const int hi = hi0 + h_r[half];                                    \
        const int wi = wi0 + h_s[half];                                    \
        const bool ok = mok && h_ok[half]                                  \
            && (unsigned)hi < (unsigned)H && (unsigned)wi < (unsigned)W;   \
        const ushort_t* px =                                               \
            ok ? X + ((((int64_t)xn * H + hi) * W + wi) * Ci               \
                      + h_ci0[half])                                       \
               : X;                                                        \
        const bf16x8w v = *reinterpret_cast<const bf16x8w*>(px);           \
        const bf16x8w z = (bf16x8w){0, 0, 0, 0, 0, 0, 0, 0};               \
        rb[half] = ok ? v : z;                                             \
      }                                                                    \
    }                                                                      \
  } while (0)

#define CW_WRITE(buf)                                                      \
  do {                                                                     \
    _Pragma("unroll") for (int half = 0; half < 2; ++half) {               \
      const int c8 = (cgrp + half * 8) * 8;                                \
      const ushort_t* ea = reinterpret_cast<const ushort_t*>(&ra[half]);   \
      const ushort_t* eb = reinterpret_cast<const ushort_t*>(&rb[half]);   \
      _Pragma("unroll") for (int j = 0; j < 8; ++j) {                      \
        As[buf][(c8 + j) * WLDS_STRIDE + mrow] = ea[j];                    \
        Bs[buf][(c8 + j) * WLDS_STRIDE + mrow] = eb[j];                    \
      }                                                                    \
    }                                                                      \
  } while (0)

    const int KTm = (int)((m1 - m0 + 31) / 32);
    CW_LOAD(m0);
    CW_WRITE(0);
    for (int kt = 0; kt < KTm; ++kt) {
      __syncthreads();
      const int buf = kt & 1;
      if (kt + 1 < KTm) CW_LOAD(m0 + (int64_t)(kt + 1) * 32);

      bf16x8w afrag[4], bfrag[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        afrag[i] = *reinterpret_cast<const bf16x8w*>(
            As[buf] + (wm + i * 16 + frow) * WLDS_STRIDE + fk0);
        bfrag[i] = *reinterpret_cast<const bf16x8w*>(
            Bs[buf] + (wn + i * 16 + frow) * WLDS_STRIDE + fk0);
      }
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[i], bfrag[j], acc[i][j], 0, 0, 0);

      __syncthreads();
      if (kt + 1 < KTm) CW_WRITE(buf ^ 1);
    }
#undef CW_LOAD
#undef CW_WRITE

    float* out = partials + (int64_t)s * Co * K9;
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j)
#pragma unroll
        for (int rr = 0; rr < 4; ++rr) {
          const int gco = tco + wm + i * 16 + (lane >> 4) * 4 + rr;
          const int gcol = tcol + wn + j * 16 + (lane & 15);
          if (gco < Co && gcol < K9)
            out[(int64_t)gco * K9 + gcol] = acc[i][j][rr];
        }
  }
}

__global__ void k_conv3x3_wgrad_reduce(const float* __restrict__ P,
                                       float* __restrict__ dw,
                                       int64_t numel, int split) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < numel; i += stride) {
    float acc = 0.f;
    for (int k = 0; k < split; ++k) acc += P[(int64_t)k * numel + i];
    dw[i] = acc;
  }
}

}  // namespace

extern "C" {

void sgp_conv3x3_nhwc_bf16(const ushort_t* X, const ushort_t* Wt,
                           ushort_t* Y, float* P, int Nb, int H, int W,
                           int Ci, int Co, int Ho, int Wo, int stride,
                           int split, hipStream_t s) {
  const int64_t M = (int64_t)Nb * Ho * Wo;
  const int64_t tiles =
      ((M + CBM - 1) / CBM) * (int64_t)((Co + CBN - 1) / CBN);
  if (split > 1) {
    const int grid = conv_grid(tiles * split);
    if (stride == 1)
      hipLaunchKernelGGL((k_conv3x3_nhwc_bf16<1, true>), dim3(grid),
                         dim3(256), 0, s, X, Wt, Y, P, Nb, H, W, Ci, Co,
                         Ho, Wo, split);
    else
      hipLaunchKernelGGL((k_conv3x3_nhwc_bf16<2, true>), dim3(grid),
                         dim3(256), 0, s, X, Wt, Y, P, Nb, H, W, Ci, Co,
                         Ho, Wo, split);
    const int64_t numel = M * Co;
    int rgrid = (int)(((numel + 255) / 256) > 8192 ? 8192
                                                   : (numel + 255) / 256);
    hipLaunchKernelGGL(k_conv3x3_reduce, dim3(rgrid < 1 ? 1 : rgrid),
                       dim3(256), 0, s, P, Y, numel, split);
  } else {
    const int grid = conv_grid(tiles);
    if (stride == 1)
      hipLaunchKernelGGL((k_conv3x3_nhwc_bf16<1, false>), dim3(grid),
                         dim3(256), 0, s, X, Wt, Y, nullptr, Nb, H, W, Ci,
                         Co, Ho, Wo, 1);
    else
      hipLaunchKernelGGL((k_conv3x3_nhwc_bf16<2, false>), dim3(grid),
                         dim3(256), 0, s, X, Wt, Y, nullptr, Nb, H, W, Ci,
                         Co, Ho, Wo, 1);
  }
}

void sgp_conv3x3_wgrad_bf16(const ushort_t* dy, const ushort_t* X,
                            float* partials, float* dw, int Nb, int H,
                            int W, int Ci, int Co, int Ho, int Wo,
                            int stride, int split, hipStream_t s) {
  const int K9 = 9 * Ci;
  const int64_t tiles =
      (int64_t)((Co + 127) / 128) * ((K9 + 127) / 128) * split;
  int grid = (int)(tiles > 16384 ? 16384 : tiles);
  grid = (grid + 7) & ~7;
  if (grid < 8) grid = 8;
  if (stride == 1)
    hipLaunchKernelGGL(k_conv3x3_wgrad<1>, dim3(grid), dim3(256), 0, s,
                       dy, X, partials, Nb, H, W, Ci, Co, Ho, Wo, split);
  else
    hipLaunchKernelGGL(k_conv3x3_wgrad<2>, dim3(grid), dim3(256), 0, s,
                       dy, X, partials, Nb, H, W, Ci, Co, Ho, Wo, split);
  const int64_t numel = (int64_t)Co * K9;
  int rgrid = (int)(((numel + 255) / 256) > 8192 ? 8192
                                                 : (numel + 255) / 256);
  hipLaunchKernelGGL(k_conv3x3_wgrad_reduce, dim3(rgrid < 1 ? 1 : rgrid),
                     dim3(256), 0, s, partials, dw, numel, split);
}

}  // extern "C"
