// Fused flat-buffer kernels for the gossip hot path — gfx950 (CDNA4).
//
// Every kernel is a single launch over the whole flat parameter buffer
// (~25.6 M fp32 for ResNet-50), replacing the reference's ~161 per-tensor
// elementwise CUDA ops per step (reference gossip/distributed.py:298-314,
// 372-379, 402-425; gossip/ad_psgd.py:340-341, 357-361).
//
// Design notes (per the CDNA4 programming guide):
// * memory-bound elementwise: vectorize to float4 (16 B/lane), grid-stride
//   loop, 256-thread blocks (4 waves of 64) — targets the ~6.3 TB/s HBM3E
//   ceiling.
// * push-sum scalars (ps_weight, mixing factor) are passed as 1-element
//   device tensors and read by pointer inside the kernel, so the training
//   loop never synchronizes device->host for a scalar.
// * tails handled in-kernel (no second launch).

#include <hip/hip_runtime.h>

#include <cstdint>

#define THREADS 256

namespace {

__device__ __forceinline__ float4 ld4(const float* p) {
  return *reinterpret_cast<const float4*>(p);
}
__device__ __forceinline__ void st4(float* p, float4 v) {
  *reinterpret_cast<float4*>(p) = v;
}

// ---------------------------------------------------------------- scale_
// x *= *a
__global__ void k_scale(float* __restrict__ x, const float* __restrict__ a,
                        int64_t n4, int64_t n) {
  const float s = *a;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < n4; i += stride) {
    float4 v = ld4(x + 4 * i);
    v.x *= s; v.y *= s; v.z *= s; v.w *= s;
    st4(x + 4 * i, v);
  }
  // tail
  for (int64_t j = 4 * n4 + (blockIdx.x * blockDim.x + threadIdx.x);
       j < n; j += stride)
    x[j] *= s;
}

// ------------------------------------------------------------ add_scale_
// x = (x + r) * *a
__global__ void k_add_scale(float* __restrict__ x,
                            const float* __restrict__ r,
                            const float* __restrict__ a,
                            int64_t n4, int64_t n) {
  const float s = *a;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < n4; i += stride) {
    float4 v = ld4(x + 4 * i);
    float4 u = ld4(r + 4 * i);
    v.x = (v.x + u.x) * s; v.y = (v.y + u.y) * s;
    v.z = (v.z + u.z) * s; v.w = (v.w + u.w) * s;
    st4(x + 4 * i, v);
  }
  for (int64_t j = 4 * n4 + (blockIdx.x * blockDim.x + threadIdx.x);
       j < n; j += stride)
    x[j] = (x[j] + r[j]) * s;
}

// ------------------------------------------------------------- pack_mix_
// t = x * *a ; x = t ; out = t   (one read, two writes)
__global__ void k_pack_mix(float* __restrict__ x, float* __restrict__ out,
                           const float* __restrict__ a,
                           int64_t n4, int64_t n) {
  const float s = *a;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < n4; i += stride) {
    float4 v = ld4(x + 4 * i);
    v.x *= s; v.y *= s; v.z *= s; v.w *= s;
    st4(x + 4 * i, v);
    st4(out + 4 * i, v);
  }
  for (int64_t j = 4 * n4 + (blockIdx.x * blockDim.x + threadIdx.x);
       j < n; j += stride) {
    const float t = x[j] * s;
    x[j] = t;
    out[j] = t;
  }
}

// -------------------------------------------------------------- average_
// x = (x + y) * 0.5
__global__ void k_average(float* __restrict__ x, const float* __restrict__ y,
                          int64_t n4, int64_t n) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < n4; i += stride) {
    float4 v = ld4(x + 4 * i);
    float4 u = ld4(y + 4 * i);
    v.x = (v.x + u.x) * 0.5f; v.y = (v.y + u.y) * 0.5f;
    v.z = (v.z + u.z) * 0.5f; v.w = (v.w + u.w) * 0.5f;
    st4(x + 4 * i, v);
  }
  for (int64_t j = 4 * n4 + (blockIdx.x * blockDim.x + threadIdx.x);
       j < n; j += stride)
    x[j] = (x[j] + y[j]) * 0.5f;
}

// -------------------------------------------- bf16 wire-format variants
// Gossip messages can travel as bf16 (halves xGMI bytes per exchange);
// master params stay fp32.  Conversion is RNE, fused into the pack /
// accumulate pass.

__device__ __forceinline__ unsigned short f2b_(float f) {
  union { unsigned int i; float f; } v;
  v.f = f;
  unsigned int r = v.i + 0x7FFFu + ((v.i >> 16) & 1u);
  return (unsigned short)(r >> 16);
}

__device__ __forceinline__ float b2f_(unsigned short u) {
  union { unsigned int i; float f; } v;
  v.i = ((unsigned int)u) << 16;
  return v.f;
}

// t = x * *a ; x = t ; out_bf16 = bf16(t)
__global__ void k_pack_mix_bf16(float* __restrict__ x,
                                unsigned short* __restrict__ out,
                                const float* __restrict__ a,
                                int64_t n4, int64_t n) {
  const float s = *a;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < n4; i += stride) {
    float4 v = ld4(x + 4 * i);
    v.x *= s; v.y *= s; v.z *= s; v.w *= s;
    st4(x + 4 * i, v);
    ushort4 o;
    o.x = f2b_(v.x); o.y = f2b_(v.y); o.z = f2b_(v.z); o.w = f2b_(v.w);
    *reinterpret_cast<ushort4*>(out + 4 * i) = o;
  }
  for (int64_t j = 4 * n4 + (blockIdx.x * blockDim.x + threadIdx.x);
       j < n; j += stride) {
    const float t = x[j] * s;
    x[j] = t;
    out[j] = f2b_(t);
  }
}

// x = (x + float(r_bf16)) * *a
__global__ void k_add_scale_bf16(float* __restrict__ x,
                                 const unsigned short* __restrict__ r,
                                 const float* __restrict__ a,
                                 int64_t n4, int64_t n) {
  const float s = *a;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < n4; i += stride) {
    float4 v = ld4(x + 4 * i);
    ushort4 u = *reinterpret_cast<const ushort4*>(r + 4 * i);
    v.x = (v.x + b2f_(u.x)) * s;
    v.y = (v.y + b2f_(u.y)) * s;
    v.z = (v.z + b2f_(u.z)) * s;
    v.w = (v.w + b2f_(u.w)) * s;
    st4(x + 4 * i, v);
  }
  for (int64_t j = 4 * n4 + (blockIdx.x * blockDim.x + threadIdx.x);
       j < n; j += stride)
    x[j] = (x[j] + b2f_(r[j])) * s;
}

// ------------------------------------------------------------- sgd_step_
// torch.optim.SGD semantics over flat buffers, one pass:
//   d   = g + wd * p
//   buf = first ? d : mu * buf + (1 - damp) * d
//   d   = nesterov ? d + mu * buf : buf        (mu != 0)
//   p  -= lr * d
template <bool kMomentum, bool kNesterov, bool kFirst>
__global__ void k_sgd(float* __restrict__ p, const float* __restrict__ g,
                      float* __restrict__ buf,
                      const float* __restrict__ lr_ptr, float mu, float wd,
                      float damp, int64_t n4, int64_t n) {
  const float lr = *lr_ptr;  // device-resident: live value under hipGraph replay
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < n4; i += stride) {
    float4 pv = ld4(p + 4 * i);
    float4 gv = ld4(g + 4 * i);
    float d0 = gv.x + wd * pv.x, d1 = gv.y + wd * pv.y,
          d2 = gv.z + wd * pv.z, d3 = gv.w + wd * pv.w;
    if (kMomentum) {
      float4 bv;
      if (kFirst) {
        bv = make_float4(d0, d1, d2, d3);
      } else {
        bv = ld4(buf + 4 * i);
        bv.x = mu * bv.x + (1.f - damp) * d0;
        bv.y = mu * bv.y + (1.f - damp) * d1;
        bv.z = mu * bv.z + (1.f - damp) * d2;
        bv.w = mu * bv.w + (1.f - damp) * d3;
      }
      st4(buf + 4 * i, bv);
      if (kNesterov) {
        d0 += mu * bv.x; d1 += mu * bv.y; d2 += mu * bv.z; d3 += mu * bv.w;
      } else {
        d0 = bv.x; d1 = bv.y; d2 = bv.z; d3 = bv.w;
      }
    }
    pv.x -= lr * d0; pv.y -= lr * d1; pv.z -= lr * d2; pv.w -= lr * d3;
    st4(p + 4 * i, pv);
  }
  for (int64_t j = 4 * n4 + (blockIdx.x * blockDim.x + threadIdx.x);
       j < n; j += stride) {
    float d = g[j] + wd * p[j];
    if (kMomentum) {
      float b = kFirst ? d : mu * buf[j] + (1.f - damp) * d;
      buf[j] = b;
      d = kNesterov ? d + mu * b : b;
    }
    p[j] -= lr * d;
  }
}

// SGD with bf16 gradients and a bf16 working-weight shadow: the fp32
// master is updated from bf16 grads (mixed-precision recipe) and the
// bf16 copy the model computes with is refreshed in the same pass —
// removes every per-layer autocast weight-cast and grad-cast kernel
// from the step.
template <bool kMomentum, bool kNesterov, bool kFirst>
__global__ void k_sgd_bf16gs(float* __restrict__ p,
                             const unsigned short* __restrict__ g,
                             float* __restrict__ buf,
                             unsigned short* __restrict__ shadow,
                             const float* __restrict__ lr_ptr, float mu,
                             float wd, float damp, int64_t n4, int64_t n) {
  const float lr = *lr_ptr;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < n4; i += stride) {
    float4 pv = ld4(p + 4 * i);
    const ushort4 gu = *reinterpret_cast<const ushort4*>(g + 4 * i);
    float d0 = b2f_(gu.x) + wd * pv.x, d1 = b2f_(gu.y) + wd * pv.y,
          d2 = b2f_(gu.z) + wd * pv.z, d3 = b2f_(gu.w) + wd * pv.w;
    if (kMomentum) {
      float4 bv;
      if (kFirst) {
        bv = make_float4(d0, d1, d2, d3);
      } else {
        bv = ld4(buf + 4 * i);
        bv.x = mu * bv.x + (1.f - damp) * d0;
        bv.y = mu * bv.y + (1.f - damp) * d1;
        bv.z = mu * bv.z + (1.f - damp) * d2;
        bv.w = mu * bv.w + (1.f - damp) * d3;
      }
      st4(buf + 4 * i, bv);
      if (kNesterov) {
        d0 += mu * bv.x; d1 += mu * bv.y; d2 += mu * bv.z; d3 += mu * bv.w;
      } else {
        d0 = bv.x; d1 = bv.y; d2 = bv.z; d3 = bv.w;
      }
    }
    pv.x -= lr * d0; pv.y -= lr * d1; pv.z -= lr * d2; pv.w -= lr * d3;
    st4(p + 4 * i, pv);
    ushort4 sv;
    sv.x = f2b_(pv.x); sv.y = f2b_(pv.y);
    sv.z = f2b_(pv.z); sv.w = f2b_(pv.w);
    *reinterpret_cast<ushort4*>(shadow + 4 * i) = sv;
  }
  for (int64_t j = 4 * n4 + (blockIdx.x * blockDim.x + threadIdx.x);
       j < n; j += stride) {
    float d = b2f_(g[j]) + wd * p[j];
    if (kMomentum) {
      float b = kFirst ? d : mu * buf[j] + (1.f - damp) * d;
      buf[j] = b;
      d = kNesterov ? d + mu * b : b;
    }
    p[j] -= lr * d;
    shadow[j] = f2b_(p[j]);
  }
}

// fp32 -> bf16 shadow refresh (after gossip merges touch the master)
__global__ void k_cast_shadow(const float* __restrict__ p,
                              unsigned short* __restrict__ shadow,
                              int64_t n4, int64_t n) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < n4; i += stride) {
    const float4 pv = ld4(p + 4 * i);
    ushort4 sv;
    sv.x = f2b_(pv.x); sv.y = f2b_(pv.y);
    sv.z = f2b_(pv.z); sv.w = f2b_(pv.w);
    *reinterpret_cast<ushort4*>(shadow + 4 * i) = sv;
  }
  for (int64_t j = 4 * n4 + (blockIdx.x * blockDim.x + threadIdx.x);
       j < n; j += stride)
    shadow[j] = f2b_(p[j]);
}

// Multi-tensor gather: concatenate nparams scattered gradient tensors
// (device pointer table) into the flat gradient buffer in ONE launch.
// Replaces autograd's per-parameter accumulate adds (~161 tiny kernels
// per step) under steal-mode grads: autograd ASSIGNS fresh tensors
// (no kernel), this gathers them.  offsets[i] is the flat start of
// param i, offsets[nparams] = total.  Each thread binary-searches its
// flat index into a param (log2(161) ~ 8 steps, amortized over a
// grid-stride loop).  Templated on element type (bf16 shadow-grad
// section vs fp32).
// Each thread handles an 8-element granule: ONE search per granule
// (offsets staged in LDS — 8 dependent global loads per element was
// latency-bound at 154 us for the 25.6 M-element ResNet-50 gather),
// and the common in-param granule copies as 4 x 4 B when the source is
// 4 B co-aligned.
template <typename T>
__global__ void k_gather_multi(const T* const* __restrict__ srcs,
                               const int64_t* __restrict__ offsets,
                               T* __restrict__ out, int nparams,
                               int64_t total) {
  extern __shared__ int64_t soff[];  // [nparams + 1]
  for (int p = threadIdx.x; p <= nparams; p += blockDim.x)
    soff[p] = offsets[p];
  __syncthreads();

  const int64_t g_total = (total + 7) >> 3;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t g = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       g < g_total; g += stride) {
    const int64_t i0 = g << 3;
    const int64_t i1 = min(i0 + 8, total);
    int lo = 0, hi = nparams - 1;
    while (lo < hi) {
      const int mid = (lo + hi + 1) >> 1;
      if (soff[mid] <= i0) lo = mid;
      else hi = mid - 1;
    }
    const T* src = srcs[lo];
    const int64_t off = soff[lo];
    const int64_t end = soff[lo + 1];
    if (i1 <= end) {
      // whole granule inside one param
      if (src == nullptr) {
        for (int64_t i = i0; i < i1; ++i) out[i] = (T)0;
      } else if (i1 - i0 == 8 && (((i0 - off) * sizeof(T)) & 3) == 0 &&
                 ((uintptr_t)(src + (i0 - off)) & 3) == 0) {
        const unsigned int* s =
            reinterpret_cast<const unsigned int*>(src + (i0 - off));
        unsigned int* d = reinterpret_cast<unsigned int*>(out + i0);
        const int words = 8 * sizeof(T) / 4;
#pragma unroll
        for (int wdx = 0; wdx < words; ++wdx) d[wdx] = s[wdx];
      } else {
        for (int64_t i = i0; i < i1; ++i) out[i] = src[i - off];
      }
    } else {
      // granule straddles a param boundary: per-element path
      for (int64_t i = i0; i < i1; ++i) {
        int l2 = lo;
        while (l2 + 1 <= nparams - 1 && soff[l2 + 1] <= i) ++l2;
        const T* s2 = srcs[l2];
        out[i] = s2 ? s2[i - soff[l2]] : (T)0;
      }
    }
  }
}

inline int grid_for(int64_t work) {
  int64_t blocks = (work + THREADS - 1) / THREADS;
  if (blocks < 1) blocks = 1;
  if (blocks > 1048576) blocks = 1048576;
  return (int)blocks;
}

}  // namespace

// -------------------------------------------------------------- C wrappers
// (launched on the stream passed by the bindings; stream is the caller's
// current torch stream so ops are ordered with the training stream)

extern "C" {

void sgp_scale(float* x, const float* a, int64_t n, hipStream_t stream) {
  const int64_t n4 = n / 4;
  hipLaunchKernelGGL(k_scale, dim3(grid_for(n4 ? n4 : n)), dim3(THREADS), 0,
                     stream, x, a, n4, n);
}

void sgp_add_scale(float* x, const float* r, const float* a, int64_t n,
                   hipStream_t stream) {
  const int64_t n4 = n / 4;
  hipLaunchKernelGGL(k_add_scale, dim3(grid_for(n4 ? n4 : n)), dim3(THREADS),
                     0, stream, x, r, a, n4, n);
}

void sgp_pack_mix(float* x, float* out, const float* a, int64_t n,
                  hipStream_t stream) {
  const int64_t n4 = n / 4;
  hipLaunchKernelGGL(k_pack_mix, dim3(grid_for(n4 ? n4 : n)), dim3(THREADS),
                     0, stream, x, out, a, n4, n);
}

void sgp_average(float* x, const float* y, int64_t n, hipStream_t stream) {
  const int64_t n4 = n / 4;
  hipLaunchKernelGGL(k_average, dim3(grid_for(n4 ? n4 : n)), dim3(THREADS), 0,
                     stream, x, y, n4, n);
}

void sgp_pack_mix_bf16(float* x, unsigned short* out, const float* a,
                       int64_t n, hipStream_t stream) {
  const int64_t n4 = n / 4;
  hipLaunchKernelGGL(k_pack_mix_bf16, dim3(grid_for(n4 ? n4 : n)),
                     dim3(THREADS), 0, stream, x, out, a, n4, n);
}

void sgp_add_scale_bf16(float* x, const unsigned short* r, const float* a,
                        int64_t n, hipStream_t stream) {
  const int64_t n4 = n / 4;
  hipLaunchKernelGGL(k_add_scale_bf16, dim3(grid_for(n4 ? n4 : n)),
                     dim3(THREADS), 0, stream, x, r, a, n4, n);
}

void sgp_sgd_step(float* p, const float* g, float* buf, const float* lr_ptr,
                  double mu, double wd, double damp, bool nesterov,
                  bool first, int64_t n, hipStream_t stream) {
  const int64_t n4 = n / 4;
  const dim3 grid(grid_for(n4 ? n4 : n));
  const float muf = (float)mu, wdf = (float)wd, dampf = (float)damp;
#define LAUNCH(M, N, F)                                                   \
  hipLaunchKernelGGL((k_sgd<M, N, F>), grid, dim3(THREADS), 0, stream, p, \
                     g, buf, lr_ptr, muf, wdf, dampf, n4, n)
  if (mu != 0.0) {
    if (nesterov) {
      if (first) LAUNCH(true, true, true);
      else LAUNCH(true, true, false);
    } else {
      if (first) LAUNCH(true, false, true);
      else LAUNCH(true, false, false);
    }
  } else {
    LAUNCH(false, false, false);
  }
#undef LAUNCH
}

void sgp_sgd_step_bf16gs(float* p, const unsigned short* g, float* buf,
                         unsigned short* shadow, const float* lr_ptr,
                         double mu, double wd, double damp, bool nesterov,
                         bool first, int64_t n, hipStream_t stream) {
  const int64_t n4 = n / 4;
  const dim3 grid(grid_for(n4 ? n4 : n));
  const float muf = (float)mu, wdf = (float)wd, dampf = (float)damp;
#define LAUNCH(M, N, F)                                                   \
  hipLaunchKernelGGL((k_sgd_bf16gs<M, N, F>), grid, dim3(THREADS), 0,     \
                     stream, p, g, buf, shadow, lr_ptr, muf, wdf, dampf,  \
                     n4, n)
  if (mu != 0.0) {
    if (nesterov) {
      if (first) LAUNCH(true, true, true);
      else LAUNCH(true, true, false);
    } else {
      if (first) LAUNCH(true, false, true);
      else LAUNCH(true, false, false);
    }
  } else {
    LAUNCH(false, false, false);
  }
#undef LAUNCH
}

void sgp_cast_shadow(const float* p, unsigned short* shadow, int64_t n,
                     hipStream_t stream) {
  const int64_t n4 = n / 4;
  hipLaunchKernelGGL(k_cast_shadow, dim3(grid_for(n4 ? n4 : n)),
                     dim3(THREADS), 0, stream, p, shadow, n4, n);
}

void sgp_gather_multi_f32(const float* const* srcs, const int64_t* offsets,
                          float* out, int nparams, int64_t total,
                          hipStream_t stream) {
  const size_t shmem = (nparams + 1) * sizeof(int64_t);
  hipLaunchKernelGGL(k_gather_multi<float>,
                     dim3(grid_for((total + 7) / 8)), dim3(THREADS), shmem,
                     stream, srcs, offsets, out, nparams, total);
}

void sgp_gather_multi_bf16(const unsigned short* const* srcs,
                           const int64_t* offsets, unsigned short* out,
                           int nparams, int64_t total, hipStream_t stream) {
  const size_t shmem = (nparams + 1) * sizeof(int64_t);
  hipLaunchKernelGGL(k_gather_multi<unsigned short>,
                     dim3(grid_for((total + 7) / 8)), dim3(THREADS), shmem,
                     stream, srcs, offsets, out, nparams, total);
}

}  // extern "C"
