// Fused NHWC BatchNorm(+residual-add)(+ReLU) for gfx950 (CDNA4), bf16
// activations / fp32 statistics.
//
// Replaces, per BN layer per step, the stack of: MIOpen spatial BN
// (3 kernels fwd + 3 bwd), autocast bf16<->fp32 casts (torch runs BN in
// fp32 under autocast), the separate ReLU/clamp kernels, the residual
// add, and the num_batches_tracked counter bump — profiled together at
// ~38% of a graphed ResNet-50 bs=32 step (profiles/r01_*).
//
// Layout: channels_last (NHWC): a [N,C,H,W] tensor is a row-major
// M x C matrix with M = N*H*W.  Per-channel statistics are column sums.
//
// Performance notes (measured on MI355X):
// * Reductions are two-stage and atomic-free: each block stores its
//   per-channel partial sums to partials[block][2C] (plain stores), and
//   the finalize kernel sums the <=512 partials.  A single-copy
//   atomicAdd version measured 400us on a 100k x 256 reduce (50x off the
//   HBM bound) from cross-block serialization on 2C addresses; an
//   8-shadow version still measured 44us.  Partial-store is also
//   deterministic (fixed summation order).
// * Apply kernels exploit that the grid stride (gridDim*256*8 elements)
//   is always a multiple of C (C = 8*2^k <= 2048 divides 2048), so each
//   lane's 8-channel group is loop-invariant: all per-channel constants
//   are hoisted into registers and the inner loop is pure streaming
//   (one 16-B load [+1 per extra stream], one 16-B store).
// * bf16 conversion is round-to-nearest-even, in-register; activations
//   stay bf16 end-to-end (no autocast fp32 round trip).

#include <hip/hip_runtime.h>

#include <cstdint>

#define THREADS 256

namespace {

typedef unsigned short ushort_t;

__device__ __forceinline__ float b2f(ushort_t u) {
  union { unsigned int i; float f; } v;
  v.i = ((unsigned int)u) << 16;
  return v.f;
}

__device__ __forceinline__ ushort_t f2b(float f) {
  union { unsigned int i; float f; } v;
  v.f = f;
  unsigned int r = v.i + 0x7FFFu + ((v.i >> 16) & 1u);
  return (ushort_t)(r >> 16);
}

struct U4 { unsigned int x, y, z, w; };  // 8 bf16

__device__ __forceinline__ U4 ld8h(const ushort_t* p) {
  return *reinterpret_cast<const U4*>(p);
}
__device__ __forceinline__ void st8h(ushort_t* p, U4 v) {
  *reinterpret_cast<U4*>(p) = v;
}

__device__ __forceinline__ void unpack8(U4 v, float* f) {
  f[0] = b2f((ushort_t)(v.x & 0xFFFF)); f[1] = b2f((ushort_t)(v.x >> 16));
  f[2] = b2f((ushort_t)(v.y & 0xFFFF)); f[3] = b2f((ushort_t)(v.y >> 16));
  f[4] = b2f((ushort_t)(v.z & 0xFFFF)); f[5] = b2f((ushort_t)(v.z >> 16));
  f[6] = b2f((ushort_t)(v.w & 0xFFFF)); f[7] = b2f((ushort_t)(v.w >> 16));
}

__device__ __forceinline__ U4 pack8(const float* f) {
  U4 v;
  v.x = (unsigned int)f2b(f[0]) | ((unsigned int)f2b(f[1]) << 16);
  v.y = (unsigned int)f2b(f[2]) | ((unsigned int)f2b(f[3]) << 16);
  v.z = (unsigned int)f2b(f[4]) | ((unsigned int)f2b(f[5]) << 16);
  v.w = (unsigned int)f2b(f[6]) | ((unsigned int)f2b(f[7]) << 16);
  return v;
}

// Last-block detection for reduce+finalize fusion.  VERSION 2: the
// first attempt used __threadfence() per block (device-scope release),
// which on MI355X forces a cross-XCD L2 writeback — measured 3x step
// regression.  Here the PARTIALS THEMSELVES are written with
// agent-scope (system-coherent, sc1) stores, so no fence is needed:
// the stores bypass the non-coherent per-XCD caching and the last
// block reads them back with agent-scope loads.
__device__ __forceinline__ void st_agent(float* p, float v) {
  __hip_atomic_store(p, v, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
}
__device__ __forceinline__ float ld_agent(const float* p) {
  return __hip_atomic_load(p, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
}

__device__ __forceinline__ bool bn_signal_last(unsigned int* ctr) {
  __syncthreads();  // all of this block's partial stores issued
  __shared__ bool last;
  if (threadIdx.x == 0) {
    const unsigned int prev = __hip_atomic_fetch_add(
        ctr, 1u, __ATOMIC_ACQ_REL, __HIP_MEMORY_SCOPE_AGENT);
    last = (prev == gridDim.x - 1);
  }
  __syncthreads();
  return last;
}

// One-block finalize bodies (the standalone finalize kernels' math run
// by 256 threads looping channel groups of 8 x 32 slices).
__device__ void bn_fwd_finalize_block(
    const float* __restrict__ partials, const float* __restrict__ gamma,
    const float* __restrict__ beta, float* __restrict__ running_mean,
    float* __restrict__ running_var, float* __restrict__ save_mean,
    float* __restrict__ save_invstd, float* __restrict__ scale_shift,
    float momentum, float eps, int64_t M, int C, int nblocks,
    bool update_running) {
  __shared__ float red[256 * 2];
  const int cl = threadIdx.x % 8;
  const int w = threadIdx.x / 8;
  for (int cg = 0; cg < C; cg += 8) {
    const int c = cg + cl;
    float s = 0.f, q = 0.f;
    if (c < C) {
      for (int k = w; k < nblocks; k += 32) {
        s += ld_agent(partials + (int64_t)k * 2 * C + c);
        q += ld_agent(partials + (int64_t)k * 2 * C + C + c);
      }
    }
    red[threadIdx.x] = s;
    red[256 + threadIdx.x] = q;
    __syncthreads();
    if (w == 0 && c < C) {
      for (int t = 1; t < 32; ++t) {
        s += red[t * 8 + cl];
        q += red[256 + t * 8 + cl];
      }
      const float inv_m = 1.0f / (float)M;
      const float mean = s * inv_m;
      float var = q * inv_m - mean * mean;
      if (var < 0.f) var = 0.f;
      const float invstd = rsqrtf(var + eps);
      save_mean[c] = mean;
      save_invstd[c] = invstd;
      if (update_running) {
        const float unbiased =
            (M > 1) ? var * (float)M / (float)(M - 1) : var;
        running_mean[c] += momentum * (mean - running_mean[c]);
        running_var[c] += momentum * (unbiased - running_var[c]);
      }
      const float sc = gamma[c] * invstd;
      scale_shift[c] = sc;
      scale_shift[C + c] = beta[c] - mean * sc;
    }
    __syncthreads();
  }
}

__device__ void bn_bwd_finalize_block(
    const float* __restrict__ partials, const float* __restrict__ gamma,
    const float* __restrict__ save_mean,
    const float* __restrict__ save_invstd, float* __restrict__ dgamma,
    float* __restrict__ dbeta, float* __restrict__ coef, int64_t M, int C,
    int nblocks, bool training) {
  __shared__ float red[256 * 2];
  const int cl = threadIdx.x % 8;
  const int w = threadIdx.x / 8;
  for (int cg = 0; cg < C; cg += 8) {
    const int c = cg + cl;
    float dg = 0.f, db = 0.f;
    if (c < C) {
      for (int k = w; k < nblocks; k += 32) {
        dg += ld_agent(partials + (int64_t)k * 2 * C + c);
        db += ld_agent(partials + (int64_t)k * 2 * C + C + c);
      }
    }
    red[threadIdx.x] = dg;
    red[256 + threadIdx.x] = db;
    __syncthreads();
    if (w == 0 && c < C) {
      for (int t = 1; t < 32; ++t) {
        dg += red[t * 8 + cl];
        db += red[256 + t * 8 + cl];
      }
      dgamma[c] = dg;
      dbeta[c] = db;
      const float istd = save_invstd[c];
      const float A = gamma[c] * istd;
      float D = 0.f, B = 0.f;
      if (training) {
        const float inv_m = 1.0f / (float)M;
        D = -istd * A * dg * inv_m;
        B = -A * db * inv_m - D * save_mean[c];
      }
      coef[c] = A;
      coef[C + c] = D;
      coef[2 * C + c] = B;
    }
    __syncthreads();
  }
}

// ------------------------------------------------------------ fwd reduce
// shadow[blockIdx%8][0..C) += col-sums of x ; [C..2C) += col-sums of x^2
// FUSED: the last block to finish also runs the finalize (one launch
// instead of two per BN layer).
template <bool FUSED>
__global__ void k_bn_fwd_reduce(const ushort_t* __restrict__ x,
                                float* __restrict__ partials,
                                int64_t M, int C, unsigned int* counter,
                                const float* __restrict__ gamma,
                                const float* __restrict__ beta,
                                float* __restrict__ running_mean,
                                float* __restrict__ running_var,
                                float* __restrict__ save_mean,
                                float* __restrict__ save_invstd,
                                float* __restrict__ scale_shift,
                                float momentum, float eps,
                                bool update_running) {
  const int TX = C >> 3;
  const int tx = threadIdx.x % TX;
  const int ty = threadIdx.x / TX;
  const int TY = blockDim.x / TX;
  const int c0 = tx << 3;

  float s[8] = {0}, q[8] = {0}, f[8];
  const int64_t row0 = (int64_t)blockIdx.x * TY + ty;
  const int64_t rstride = (int64_t)gridDim.x * TY;
  for (int64_t r = row0; r < M; r += rstride) {
    unpack8(ld8h(x + r * C + c0), f);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      s[j] += f[j];
      q[j] += f[j] * f[j];
    }
  }

  __shared__ float lds[THREADS * 16];
  float* mine = lds + threadIdx.x * 16;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    mine[j] = s[j];
    mine[8 + j] = q[j];
  }
  __syncthreads();
  if (ty == 0) {
    for (int t = 1; t < TY; ++t) {
      const float* other = lds + (t * TX + tx) * 16;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        s[j] += other[j];
        q[j] += other[8 + j];
      }
    }
    float* part = partials + (int64_t)blockIdx.x * 2 * C;
    if (FUSED) {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        st_agent(part + c0 + j, s[j]);
        st_agent(part + C + c0 + j, q[j]);
      }
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        part[c0 + j] = s[j];
        part[C + c0 + j] = q[j];
      }
    }
  }
  if (FUSED) {
    if (bn_signal_last(counter)) {
      bn_fwd_finalize_block(partials, gamma, beta, running_mean,
                            running_var, save_mean, save_invstd,
                            scale_shift, momentum, eps, M, C,
                            (int)gridDim.x, update_running);
      __syncthreads();
      if (threadIdx.x == 0) *counter = 0;
    }
  }
}

// ---------------------------------------------------------- fwd finalize
// Collapse shadows; mean/invstd; running-stat update; scale/shift pair.
// Geometry: one block per 32 channels; 8 slices sum the <=256 partials
// in parallel (coalesced over channels), LDS-reduced.  A naive
// C-threads-total version was latency-bound at ~120us (partials spread
// over 8 XCD L2s, one CU summing serially) — 60% of a fused step.
__global__ void k_bn_fwd_finalize(const float* __restrict__ partials,
                                  const float* __restrict__ gamma,
                                  const float* __restrict__ beta,
                                  float* __restrict__ running_mean,
                                  float* __restrict__ running_var,
                                  float* __restrict__ save_mean,
                                  float* __restrict__ save_invstd,
                                  float* __restrict__ scale_shift,
                                  float momentum, float eps, int64_t M,
                                  int C, int nblocks, bool update_running) {
  const int cl = threadIdx.x % 8;
  const int w = threadIdx.x / 8;  // 32 slices
  const int c = blockIdx.x * 8 + cl;
  float s = 0.f, q = 0.f;
  if (c < C) {
    for (int k = w; k < nblocks; k += 32) {
      s += partials[(int64_t)k * 2 * C + c];
      q += partials[(int64_t)k * 2 * C + C + c];
    }
  }
  __shared__ float red[256 * 2];
  red[threadIdx.x] = s;
  red[256 + threadIdx.x] = q;
  __syncthreads();
  if (w != 0 || c >= C) return;
  for (int t = 1; t < 32; ++t) {
    s += red[t * 8 + cl];
    q += red[256 + t * 8 + cl];
  }
  const float inv_m = 1.0f / (float)M;
  const float mean = s * inv_m;
  float var = q * inv_m - mean * mean;
  if (var < 0.f) var = 0.f;
  const float invstd = rsqrtf(var + eps);
  save_mean[c] = mean;
  save_invstd[c] = invstd;
  if (update_running) {
    const float unbiased = (M > 1) ? var * (float)M / (float)(M - 1) : var;
    running_mean[c] += momentum * (mean - running_mean[c]);
    running_var[c] += momentum * (unbiased - running_var[c]);
  }
  const float sc = gamma[c] * invstd;
  scale_shift[c] = sc;
  scale_shift[C + c] = beta[c] - mean * sc;
}

// ------------------------------------------------------------- eval prep
__global__ void k_bn_eval_prep(const float* __restrict__ running_mean,
                               const float* __restrict__ running_var,
                               const float* __restrict__ gamma,
                               const float* __restrict__ beta,
                               float* __restrict__ scale_shift, float eps,
                               int C) {
  const int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  const float invstd = rsqrtf(running_var[c] + eps);
  const float sc = gamma[c] * invstd;
  scale_shift[c] = sc;
  scale_shift[C + c] = beta[c] - running_mean[c] * sc;
}

// ------------------------------------------------------------- fwd apply
// y = [relu]( x*scale + shift [+ res] ) — per-channel constants hoisted
template <bool kRelu, bool kRes>
__global__ void k_bn_fwd_apply(const ushort_t* __restrict__ x,
                               const ushort_t* __restrict__ res,
                               ushort_t* __restrict__ y,
                               const float* __restrict__ scale_shift,
                               int64_t total8, int C) {
  const int64_t i0 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  // stride*8 % C == 0 -> channel group is loop-invariant
  const int c0 = (int)((i0 << 3) % C);
  float sc[8], sh[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    sc[j] = scale_shift[c0 + j];
    sh[j] = scale_shift[C + c0 + j];
  }
  float f[8], rz[8];
  for (int64_t i = i0; i < total8; i += stride) {
    const int64_t e0 = i << 3;
    unpack8(ld8h(x + e0), f);
    if (kRes) unpack8(ld8h(res + e0), rz);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float t = f[j] * sc[j] + sh[j];
      if (kRes) t += rz[j];
      if (kRelu) t = t > 0.f ? t : 0.f;
      f[j] = t;
    }
    st8h(y + e0, pack8(f));
  }
}

// ------------------------------------------------------------ bwd reduce
// shadow[b%8][0..C) += sum(dy_eff * xhat) ; [C..2C) += sum(dy_eff)
// FUSED: last block also finalizes dgamma/dbeta + dx coefficients.
template <bool kRelu, bool FUSED>
__global__ void k_bn_bwd_reduce(const ushort_t* __restrict__ x,
                                const ushort_t* __restrict__ dy,
                                const ushort_t* __restrict__ y,
                                const float* __restrict__ save_mean,
                                const float* __restrict__ save_invstd,
                                float* __restrict__ partials, int64_t M,
                                int C, unsigned int* counter,
                                const float* __restrict__ gamma,
                                float* __restrict__ dgamma,
                                float* __restrict__ dbeta,
                                float* __restrict__ coef, bool training) {
  const int TX = C >> 3;
  const int tx = threadIdx.x % TX;
  const int ty = threadIdx.x / TX;
  const int TY = blockDim.x / TX;
  const int c0 = tx << 3;

  float mean[8], istd[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    mean[j] = save_mean[c0 + j];
    istd[j] = save_invstd[c0 + j];
  }

  float dg[8] = {0}, db[8] = {0}, fx[8], fdy[8], fy[8];
  const int64_t row0 = (int64_t)blockIdx.x * TY + ty;
  const int64_t rstride = (int64_t)gridDim.x * TY;
  for (int64_t r = row0; r < M; r += rstride) {
    const int64_t off = r * C + c0;
    unpack8(ld8h(x + off), fx);
    unpack8(ld8h(dy + off), fdy);
    if (kRelu) unpack8(ld8h(y + off), fy);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float d = fdy[j];
      if (kRelu && !(fy[j] > 0.f)) d = 0.f;
      db[j] += d;
      dg[j] += d * (fx[j] - mean[j]) * istd[j];
    }
  }

  __shared__ float lds[THREADS * 16];
  float* mine = lds + threadIdx.x * 16;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    mine[j] = dg[j];
    mine[8 + j] = db[j];
  }
  __syncthreads();
  if (ty == 0) {
    for (int t = 1; t < TY; ++t) {
      const float* other = lds + (t * TX + tx) * 16;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        dg[j] += other[j];
        db[j] += other[8 + j];
      }
    }
    float* part = partials + (int64_t)blockIdx.x * 2 * C;
    if (FUSED) {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        st_agent(part + c0 + j, dg[j]);
        st_agent(part + C + c0 + j, db[j]);
      }
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        part[c0 + j] = dg[j];
        part[C + c0 + j] = db[j];
      }
    }
  }
  if (FUSED) {
    if (bn_signal_last(counter)) {
      bn_bwd_finalize_block(partials, gamma, save_mean, save_invstd,
                            dgamma, dbeta, coef, M, C, (int)gridDim.x,
                            training);
      __syncthreads();
      if (threadIdx.x == 0) *counter = 0;
    }
  }
}

// ---------------------------------------------------------- bwd finalize
// Collapse shadows -> dgamma/dbeta, and the 3 per-channel coefficients of
// the affine form dx = A*dy_eff + D*x + B:
//   A = gamma*invstd
//   D = -invstd*A*dgamma/M
//   B = -A*dbeta/M + mean*invstd*A*dgamma/M  ( = -A*k2 - D*(-mean)... )
// (training; eval uses A only with D=B=0)
__global__ void k_bn_bwd_finalize(const float* __restrict__ partials,
                                  const float* __restrict__ gamma,
                                  const float* __restrict__ save_mean,
                                  const float* __restrict__ save_invstd,
                                  float* __restrict__ dgamma,
                                  float* __restrict__ dbeta,
                                  float* __restrict__ coef, int64_t M,
                                  int C, int nblocks, bool training) {
  const int cl = threadIdx.x % 8;
  const int w = threadIdx.x / 8;  // 32 slices
  const int c = blockIdx.x * 8 + cl;
  float dg = 0.f, db = 0.f;
  if (c < C) {
    for (int k = w; k < nblocks; k += 32) {
      dg += partials[(int64_t)k * 2 * C + c];
      db += partials[(int64_t)k * 2 * C + C + c];
    }
  }
  __shared__ float red[256 * 2];
  red[threadIdx.x] = dg;
  red[256 + threadIdx.x] = db;
  __syncthreads();
  if (w != 0 || c >= C) return;
  for (int t = 1; t < 32; ++t) {
    dg += red[t * 8 + cl];
    db += red[256 + t * 8 + cl];
  }
  dgamma[c] = dg;
  dbeta[c] = db;
  const float istd = save_invstd[c];
  const float A = gamma[c] * istd;
  float D = 0.f, B = 0.f;
  if (training) {
    const float inv_m = 1.0f / (float)M;
    D = -istd * A * dg * inv_m;
    B = -A * db * inv_m - D * save_mean[c];
  }
  coef[c] = A;
  coef[C + c] = D;
  coef[2 * C + c] = B;
}

// ------------------------------------------------------------- bwd apply
// dx = A*dy_eff + D*x + B ; dres = dy_eff — constants hoisted per lane
template <bool kRelu, bool kRes>
__global__ void k_bn_bwd_apply(const ushort_t* __restrict__ x,
                               const ushort_t* __restrict__ dy,
                               const ushort_t* __restrict__ y,
                               ushort_t* __restrict__ dx,
                               ushort_t* __restrict__ dres,
                               const float* __restrict__ coef,
                               int64_t total8, int C) {
  const int64_t i0 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const int c0 = (int)((i0 << 3) % C);
  float A[8], D[8], B[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    A[j] = coef[c0 + j];
    D[j] = coef[C + c0 + j];
    B[j] = coef[2 * C + c0 + j];
  }
  float fx[8], fdy[8], fy[8], o[8], orz[8];
  for (int64_t i = i0; i < total8; i += stride) {
    const int64_t e0 = i << 3;
    unpack8(ld8h(x + e0), fx);
    unpack8(ld8h(dy + e0), fdy);
    if (kRelu) unpack8(ld8h(y + e0), fy);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float d = fdy[j];
      if (kRelu && !(fy[j] > 0.f)) d = 0.f;
      if (kRes) orz[j] = d;
      o[j] = A[j] * d + D[j] * fx[j] + B[j];
    }
    st8h(dx + e0, pack8(o));
    if (kRes) st8h(dres + e0, pack8(orz));
  }
}

inline int reduce_grid(int64_t M, int C) {
  const int TY = THREADS / (C >> 3);
  int64_t blocks = (M + TY - 1) / TY;
  if (blocks > 256) blocks = 256;
  if (blocks < 1) blocks = 1;
  return (int)blocks;
}

inline int apply_grid(int64_t total8) {
  int64_t blocks = (total8 + THREADS - 1) / THREADS;
  if (blocks > 4096) blocks = 4096;
  if (blocks < 1) blocks = 1;
  return (int)blocks;
}

}  // namespace

extern "C" {

int bn_reduce_nblocks(int64_t M, int C) { return reduce_grid(M, C); }

void bn_fwd_reduce(const ushort_t* x, float* partials, int64_t M, int C,
                   hipStream_t s) {
  hipLaunchKernelGGL((k_bn_fwd_reduce<false>), dim3(reduce_grid(M, C)),
                     dim3(THREADS), 0, s, x, partials, M, C, nullptr,
                     nullptr, nullptr, nullptr, nullptr, nullptr, nullptr,
                     nullptr, 0.f, 0.f, false);
}

// one launch: reduce + last-block finalize (counter must be a zeroed
// device uint32, reset to zero by the kernel for re-use)
void bn_fwd_reduce_finalize(const ushort_t* x, float* partials,
                            unsigned int* counter, const float* gamma,
                            const float* beta, float* rmean, float* rvar,
                            float* smean, float* sinvstd,
                            float* scale_shift, double momentum, double eps,
                            int64_t M, int C, bool update_running,
                            hipStream_t s) {
  hipLaunchKernelGGL((k_bn_fwd_reduce<true>), dim3(reduce_grid(M, C)),
                     dim3(THREADS), 0, s, x, partials, M, C, counter,
                     gamma, beta, rmean, rvar, smean, sinvstd, scale_shift,
                     (float)momentum, (float)eps, update_running);
}

void bn_fwd_finalize(const float* partials, const float* gamma,
                     const float* beta, float* rmean, float* rvar,
                     float* smean, float* sinvstd, float* scale_shift,
                     double momentum, double eps, int64_t M, int C,
                     bool update_running, hipStream_t s) {
  const int blocks = (C + 7) / 8;
  hipLaunchKernelGGL(k_bn_fwd_finalize, dim3(blocks), dim3(THREADS), 0, s,
                     partials, gamma, beta, rmean, rvar, smean, sinvstd,
                     scale_shift, (float)momentum, (float)eps, M, C,
                     reduce_grid(M, C), update_running);
}

void bn_eval_prep(const float* rmean, const float* rvar, const float* gamma,
                  const float* beta, float* scale_shift, double eps, int C,
                  hipStream_t s) {
  const int blocks = (C + THREADS - 1) / THREADS;
  hipLaunchKernelGGL(k_bn_eval_prep, dim3(blocks), dim3(THREADS), 0, s, rmean,
                     rvar, gamma, beta, scale_shift, (float)eps, C);
}

void bn_fwd_apply(const ushort_t* x, const ushort_t* res, ushort_t* y,
                  const float* scale_shift, int64_t M, int C, bool relu,
                  hipStream_t s) {
  const int64_t total8 = M * C / 8;
  const dim3 grid(apply_grid(total8));
#define CASE(R, Z)                                                        \
  hipLaunchKernelGGL((k_bn_fwd_apply<R, Z>), grid, dim3(THREADS), 0, s, x, \
                     res, y, scale_shift, total8, C)
  if (relu) { if (res) CASE(true, true); else CASE(true, false); }
  else      { if (res) CASE(false, true); else CASE(false, false); }
#undef CASE
}

void bn_bwd_reduce(const ushort_t* x, const ushort_t* dy, const ushort_t* y,
                   const float* smean, const float* sinvstd, float* partials,
                   int64_t M, int C, bool relu, hipStream_t s) {
  const dim3 grid(reduce_grid(M, C));
  if (relu)
    hipLaunchKernelGGL((k_bn_bwd_reduce<true, false>), grid, dim3(THREADS),
                       0, s, x, dy, y, smean, sinvstd, partials, M, C,
                       nullptr, nullptr, nullptr, nullptr, nullptr, false);
  else
    hipLaunchKernelGGL((k_bn_bwd_reduce<false, false>), grid, dim3(THREADS),
                       0, s, x, dy, y, smean, sinvstd, partials, M, C,
                       nullptr, nullptr, nullptr, nullptr, nullptr, false);
}

void bn_bwd_reduce_finalize(const ushort_t* x, const ushort_t* dy,
                            const ushort_t* y, const float* smean,
                            const float* sinvstd, float* partials,
                            unsigned int* counter, const float* gamma,
                            float* dgamma, float* dbeta, float* coef,
                            int64_t M, int C, bool relu, bool training,
                            hipStream_t s) {
  const dim3 grid(reduce_grid(M, C));
  if (relu)
    hipLaunchKernelGGL((k_bn_bwd_reduce<true, true>), grid, dim3(THREADS),
                       0, s, x, dy, y, smean, sinvstd, partials, M, C,
                       counter, gamma, dgamma, dbeta, coef, training);
  else
    hipLaunchKernelGGL((k_bn_bwd_reduce<false, true>), grid, dim3(THREADS),
                       0, s, x, dy, y, smean, sinvstd, partials, M, C,
                       counter, gamma, dgamma, dbeta, coef, training);
}

void bn_bwd_finalize(const float* partials, const float* gamma,
                     const float* smean, const float* sinvstd, float* dgamma,
                     float* dbeta, float* coef, int64_t M, int C,
                     bool training, hipStream_t s) {
  const int blocks = (C + 7) / 8;
  hipLaunchKernelGGL(k_bn_bwd_finalize, dim3(blocks), dim3(THREADS), 0, s,
                     partials, gamma, smean, sinvstd, dgamma, dbeta, coef, M,
                     C, reduce_grid(M, C), training);
}

void bn_bwd_apply(const ushort_t* x, const ushort_t* dy, const ushort_t* y,
                  ushort_t* dx, ushort_t* dres, const float* coef, int64_t M,
                  int C, bool relu, hipStream_t s) {
  const int64_t total8 = M * C / 8;
  const dim3 grid(apply_grid(total8));
#define CASE(R, Z)                                                         \
  hipLaunchKernelGGL((k_bn_bwd_apply<R, Z>), grid, dim3(THREADS), 0, s, x, \
                     dy, y, dx, dres, coef, total8, C)
  if (relu) { if (dres) CASE(true, true); else CASE(true, false); }
  else      { if (dres) CASE(false, true); else CASE(false, false); }
#undef CASE
}

}  // extern "C"
