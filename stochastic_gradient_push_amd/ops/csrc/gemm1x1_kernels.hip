// Hand-written MFMA GEMM for the 1x1-convolution shape — gfx950 (CDNA4).
//
// NT layout: C[M,N] = A[M,K] · B[N,K]^T with both operands row-major and
// K-contiguous — exactly a channels_last 1x1 conv (A = activations
// [N*H*W, Cin], B = weights [Cout, Cin]) and its dgrad (A = dy, B = W^T
// materialized).  bf16 inputs, fp32 MFMA accumulation, bf16 output.
//
// Structure (per the CDNA4 guide's canonical GEMM):
//   * 128x128 block tile, BK=32 K-step, 256 threads = 4 waves in 2x2;
//     each wave owns a 64x64 sub-tile = 4x4 fragments of
//     v_mfma_f32_16x16x32_bf16 (one MFMA consumes the whole K-step).
//   * LDS staging with +8-element row padding: the ds_read_b128 fragment
//     reads walk rows at 80 B stride = 20 banks, conflict-free across
//     each 16-lane service group.
//   * grid-stride over M-tiles so huge M (up to 401k rows at bs=32)
//     maps onto >>256 workgroups.
//
// The probe kernel empirically verifies the A/B/C fragment lane mappings
// (run once on hardware; see test_gemm1x1_gpu.py) so the layout
// assumptions are hardware-checked rather than trusted.

#include <hip/hip_runtime.h>

#include <cstdint>

namespace {

typedef unsigned short ushort_t;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

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

// --------------------------------------------------------------- probe
// One 16x16x32 MFMA from global memory using the assumed fragment
// layout:  A-frag: lane l holds A[l%16][(l/16)*8 + j]   (j = 0..7)
//          B-frag: lane l holds B[l%16][(l/16)*8 + j]   (B is [N][K])
//          C/D  : lane l, reg r -> C[(l/16)*4 + r][l%16]
__global__ void k_mfma_probe(const ushort_t* __restrict__ A,
                             const ushort_t* __restrict__ B,
                             float* __restrict__ C) {
  const int l = threadIdx.x;
  bf16x8 a, b;
  const int row = l % 16;
  const int k0 = (l / 16) * 8;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    reinterpret_cast<ushort_t*>(&a)[j] = A[row * 32 + k0 + j];
    reinterpret_cast<ushort_t*>(&b)[j] = B[row * 32 + k0 + j];
  }
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    C[((l / 16) * 4 + r) * 16 + (l % 16)] = acc[r];
  }
}

// ----------------------------------------------------------- gemm (NT)
#define BM 128
#define BN 128
#define BK 32
#define LDS_STRIDE 40  // 32 + 8 pad (bf16 elems): 80-B rows, conflict-free

__global__ __launch_bounds__(256) void k_gemm_nt_bf16(
    const ushort_t* __restrict__ A,  // [M, K]
    const ushort_t* __restrict__ B,  // [N, K]
    ushort_t* __restrict__ C,        // [M, N]
    int64_t M, int N, int K) {
  __shared__ ushort_t As[BM * LDS_STRIDE];
  __shared__ ushort_t Bs[BN * LDS_STRIDE];

  const int tid = threadIdx.x;
  const int wave = tid >> 6;        // 4 waves, 2x2
  const int lane = tid & 63;
  const int wm = (wave >> 1) * 64;  // wave row offset in tile
  const int wn = (wave & 1) * 64;   // wave col offset in tile
  const int frow = lane & 15;       // fragment row/col within 16
  const int fk0 = (lane >> 4) * 8;  // fragment k-offset (x8)

  const int n_tiles = (N + BN - 1) / BN;
  const int64_t m_tiles = (M + BM - 1) / BM;
  const int64_t total_tiles = m_tiles * n_tiles;

  for (int64_t tile = blockIdx.x; tile < total_tiles; tile += gridDim.x) {
    const int64_t tm = (tile / n_tiles) * BM;
    const int tn = (int)(tile % n_tiles) * BN;

    f32x4 acc[4][4];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j)
        acc[i][j] = (f32x4){0.f, 0.f, 0.f, 0.f};

    for (int k0 = 0; k0 < K; k0 += BK) {
      // stage A/B tiles: 128 rows x 32 cols, 16 B (8 bf16) per access,
      // 2 segments per thread per operand
      __syncthreads();
      const bool full = (tm + BM <= M) && (tn + BN <= N);
#pragma unroll
      for (int s = 0; s < 2; ++s) {
        const int seg = tid + s * 256;     // 0..511
        const int row = seg >> 2;          // /4
        const int c8 = (seg & 3) * 8;
        const int64_t gra = tm + row;
        const int grb = tn + row;
        bf16x8 va, vb;
        if (full) {
          // one 16-B vector load per operand (scalar bf16 loads are
          // ~2.5x slower; per-element guards serialize — guide traps)
          va = *reinterpret_cast<const bf16x8*>(A + gra * K + k0 + c8);
          vb = *reinterpret_cast<const bf16x8*>(
              B + (int64_t)grb * K + k0 + c8);
        } else {
          ushort_t ta[8], tb[8];
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            ta[j] = (gra < M) ? A[gra * K + k0 + c8 + j] : (ushort_t)0;
            tb[j] = (grb < N)
                ? B[(int64_t)grb * K + k0 + c8 + j] : (ushort_t)0;
          }
          va = *reinterpret_cast<bf16x8*>(ta);
          vb = *reinterpret_cast<bf16x8*>(tb);
        }
        *reinterpret_cast<bf16x8*>(As + row * LDS_STRIDE + c8) = va;
        *reinterpret_cast<bf16x8*>(Bs + row * LDS_STRIDE + c8) = vb;
      }
      __syncthreads();

      // fragments + 16 MFMA
      bf16x8 afrag[4], bfrag[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        afrag[i] = *reinterpret_cast<const bf16x8*>(
            As + (wm + i * 16 + frow) * LDS_STRIDE + fk0);
        bfrag[i] = *reinterpret_cast<const bf16x8*>(
            Bs + (wn + i * 16 + frow) * LDS_STRIDE + fk0);
      }
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[i], bfrag[j], acc[i][j], 0, 0, 0);
    }

    // epilogue: lane l, reg r -> row (l>>4)*4+r, col l&15 of each frag
#pragma unroll
    for (int i = 0; i < 4; ++i) {
#pragma unroll
      for (int j = 0; j < 4; ++j) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int64_t gr = tm + wm + i * 16 + (lane >> 4) * 4 + r;
          const int gc = tn + wn + j * 16 + (lane & 15);
          if (gr < M && gc < N) C[gr * N + gc] = f2b(acc[i][j][r]);
        }
      }
    }
  }
}

// ------------------------------------------------- gemm v2 (pipelined)
// Same geometry, register-staged double buffering (guide T14 shape):
// global loads for tile t+1 are issued right after the barrier, compute
// runs on tile t from LDS, and the staged registers are written to the
// other LDS buffer after the next barrier — global latency hides behind
// the MFMA block.
template <int BKT>
__global__ __launch_bounds__(256) void k_gemm_nt_bf16_v2(
    const ushort_t* __restrict__ A, const ushort_t* __restrict__ B,
    ushort_t* __restrict__ C, int64_t M, int N, int K) {
  constexpr int STRIDE = BKT + 8;          // +16B pad per row
  constexpr int SEGS = BM * BKT / 8 / 256; // 16B segments per thread
  __shared__ ushort_t As[2][BM * STRIDE];
  __shared__ ushort_t Bs[2][BN * STRIDE];

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int wm = (wave >> 1) * 64;
  const int wn = (wave & 1) * 64;
  const int frow = lane & 15;
  const int fk0 = (lane >> 4) * 8;

  // each thread stages SEGS segments x (A,B): seg -> (row, col8)
  constexpr int SEG_PER_ROW = BKT / 8;
  int rr[SEGS], cc[SEGS];
#pragma unroll
  for (int s = 0; s < SEGS; ++s) {
    const int seg = tid + s * 256;
    rr[s] = seg / SEG_PER_ROW;
    cc[s] = (seg % SEG_PER_ROW) * 8;
  }

  const int n_tiles = (N + BN - 1) / BN;
  const int64_t m_tiles = (M + BM - 1) / BM;
  const int64_t total_tiles = m_tiles * n_tiles;
  const int KT = K / BKT;

  for (int64_t tile = blockIdx.x; tile < total_tiles; tile += gridDim.x) {
    const int64_t tm = (tile / n_tiles) * BM;
    const int tn = (int)(tile % n_tiles) * BN;

    f32x4 acc[4][4];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j) acc[i][j] = (f32x4){0.f, 0.f, 0.f, 0.f};

    bf16x8 ra[SEGS], rb[SEGS];
    const bool full = (tm + BM <= M) && (tn + BN <= N);

#define LOAD_TILE(k0)                                                      \
  do {                                                                     \
    if (full) {                                                            \
      _Pragma("unroll") for (int s = 0; s < SEGS; ++s) {                   \
        ra[s] = *reinterpret_cast<const bf16x8*>(                          \
            A + (tm + rr[s]) * K + (k0) + cc[s]);                          \
        rb[s] = *reinterpret_cast<const bf16x8*>(                          \
            B + (int64_t)(tn + rr[s]) * K + (k0) + cc[s]);                 \
      }                                                                    \
    } else {                                                               \
      _Pragma("unroll") for (int s = 0; s < SEGS; ++s) {                   \
        const int64_t ga = tm + rr[s];                                     \
        const int gb = tn + rr[s];                                         \
        ushort_t ta[8], tb[8];                                             \
        _Pragma("unroll") for (int j = 0; j < 8; ++j) {                    \
          ta[j] = (ga < M) ? A[ga * K + (k0) + cc[s] + j] : (ushort_t)0;   \
          tb[j] = (gb < N) ? B[(int64_t)gb * K + (k0) + cc[s] + j]         \
                           : (ushort_t)0;                                  \
        }                                                                  \
        ra[s] = *reinterpret_cast<bf16x8*>(ta);                            \
        rb[s] = *reinterpret_cast<bf16x8*>(tb);                            \
      }                                                                    \
    }                                                                      \
  } while (0)

#define WRITE_TILE(buf)                                                    \
  do {                                                                     \
    _Pragma("unroll") for (int s = 0; s < SEGS; ++s) {                     \
      *reinterpret_cast<bf16x8*>(As[buf] + rr[s] * STRIDE + cc[s]) =       \
          ra[s];                                                           \
      *reinterpret_cast<bf16x8*>(Bs[buf] + rr[s] * STRIDE + cc[s]) =       \
          rb[s];                                                           \
    }                                                                      \
  } while (0)

    LOAD_TILE(0);
    WRITE_TILE(0);

    for (int kt = 0; kt < KT; ++kt) {
      __syncthreads();
      const int buf = kt & 1;
      if (kt + 1 < KT) LOAD_TILE((int64_t)(kt + 1) * BKT);  // issue early

#pragma unroll
      for (int ks = 0; ks < BKT / 32; ++ks) {
        bf16x8 afrag[4], bfrag[4];
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          afrag[i] = *reinterpret_cast<const bf16x8*>(
              As[buf] + (wm + i * 16 + frow) * STRIDE + ks * 32 + fk0);
          bfrag[i] = *reinterpret_cast<const bf16x8*>(
              Bs[buf] + (wn + i * 16 + frow) * STRIDE + ks * 32 + fk0);
        }
#pragma unroll
        for (int i = 0; i < 4; ++i)
#pragma unroll
          for (int j = 0; j < 4; ++j)
            acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                afrag[i], bfrag[j], acc[i][j], 0, 0, 0);
      }

      __syncthreads();
      if (kt + 1 < KT) WRITE_TILE(buf ^ 1);
    }
#undef LOAD_TILE
#undef WRITE_TILE

#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int64_t gr = tm + wm + i * 16 + (lane >> 4) * 4 + r;
          const int gc = tn + wn + j * 16 + (lane & 15);
          if (gr < M && gc < N) C[gr * N + gc] = f2b(acc[i][j][r]);
        }
  }
}

// --------------------------------------------- gemm v3 (glds + swizzle)
// BK=64 only.  Full tiles are staged with __builtin_amdgcn_global_load_lds
// (16-B direct-to-LDS DMA, no staging VGPRs / ds_write pass) into an
// UNPADDED [128][64] bf16 image with the st_16x32 XOR swizzle
// (byte ^= ((byte>>9)&1)<<5): glds writes are lane-linear, so the
// swizzle is applied to the per-lane GLOBAL source address, and the
// fragment ds_read_b128 uses the matching swizzled offset (guide
// recipe; linear layout would be 8-way bank-conflicted).  Tail tiles
// (M or N remainder) fall back to guarded register staging.
#define BK3 64
#define TILE_BYTES (BM * BK3 * 2)  // 16 KiB per operand

__device__ __forceinline__ int swz(int row, int cb) {
  // byte-offset swizzle within the [128][128B] tile image
  return cb ^ (((row >> 2) & 1) << 5);
}

// XCD-aware block->tile remap.  MI355X dispatches blocks round-robin
// over the 8 XCDs (xcd = blockIdx % 8), each with a private L2.  The
// tile order is n-fastest, so consecutive TILES share the same A
// m-rows; without remapping, consecutive BLOCKS land on different XCDs
// and every XCD streams its own copy of A from HBM (A is re-read
// n_tiles times -> the 128x128 kernel is HBM-bound at 512-wide shapes).
// Mapping bid -> (bid%8)*(grid/8) + bid/8 gives each XCD a contiguous
// slab of tiles, so the A rows a slab touches stay resident in that
// XCD's L2.  Requires gridDim.x % 8 == 0 (launcher guarantees).
__device__ __forceinline__ int64_t xcd_virtual_bid() {
  const int g8 = gridDim.x >> 3;
  return (int64_t)(blockIdx.x & 7) * g8 + (blockIdx.x >> 3);
}

template <bool XCDMAP>
__global__ __launch_bounds__(256) void k_gemm_nt_bf16_v3(
    const ushort_t* __restrict__ A, const ushort_t* __restrict__ B,
    ushort_t* __restrict__ C, int64_t M, int N, int K) {
  // one shared object only (a second one forces vmcnt(0) before every
  // ds_read of a glds pipeline — guide trap 4a)
  __shared__ ushort_t lds[2 * 2 * BM * BK3];  // [buf][A,B][128][64]

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int wm = (wave >> 1) * 64;
  const int wn = (wave & 1) * 64;
  const int frow = lane & 15;
  const int fkb = (lane >> 4) * 16;  // fragment k-offset in BYTES

  // glds geometry: each wave stages 4 chunks of 1 KiB per operand;
  // chunk c covers rows [c*8, c*8+8), lane l -> row c*8 + l/8,
  // byte col (l%8)*16 within the 128-B row
  const int g_row_in_chunk = lane >> 3;
  const int g_cb = (lane & 7) * 16;

  const int n_tiles = (N + BN - 1) / BN;
  const int64_t m_tiles = (M + BM - 1) / BM;
  const int64_t total_tiles = m_tiles * n_tiles;
  const int KT = K / BK3;

  const int64_t bid0 = XCDMAP ? xcd_virtual_bid() : (int64_t)blockIdx.x;
  for (int64_t tile = bid0; tile < total_tiles; tile += gridDim.x) {
    const int64_t tm = (tile / n_tiles) * BM;
    const int tn = (int)(tile % n_tiles) * BN;
    const bool full = (tm + BM <= M) && (tn + BN <= N);

    f32x4 acc[4][4];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j) acc[i][j] = (f32x4){0.f, 0.f, 0.f, 0.f};

    if (full) {
      // ---- glds pipeline (2 LDS buffers, vmcnt drained by barrier) ----
#define GLDS_TILE(buf, k0)                                                 \
  do {                                                                     \
    _Pragma("unroll") for (int c = 0; c < 4; ++c) {                        \
      const int row = (wave * 4 + c) * 8 + g_row_in_chunk;                 \
      const int scb = swz(row, g_cb);                                      \
      __builtin_amdgcn_global_load_lds(                                    \
          (const __attribute__((address_space(1))) unsigned int*)(         \
              A + (tm + row) * K + (k0) + scb / 2),                        \
          (__attribute__((address_space(3))) unsigned int*)(               \
              lds + (buf) * 2 * BM * BK3 + (wave * 4 + c) * 512),          \
          16, 0, 0);                                                       \
      __builtin_amdgcn_global_load_lds(                                    \
          (const __attribute__((address_space(1))) unsigned int*)(         \
              B + (int64_t)(tn + row) * K + (k0) + scb / 2),               \
          (__attribute__((address_space(3))) unsigned int*)(               \
              lds + ((buf) * 2 + 1) * BM * BK3 + (wave * 4 + c) * 512),    \
          16, 0, 0);                                                       \
    }                                                                      \
  } while (0)

      GLDS_TILE(0, 0);
      for (int kt = 0; kt < KT; ++kt) {
        const int buf = kt & 1;
        __syncthreads();  // drains the in-flight glds for buf (vmcnt(0))
        if (kt + 1 < KT) GLDS_TILE(buf ^ 1, (int64_t)(kt + 1) * BK3);

        const ushort_t* As_ = lds + buf * 2 * BM * BK3;
        const ushort_t* Bs_ = lds + (buf * 2 + 1) * BM * BK3;
#pragma unroll
        for (int ks = 0; ks < 2; ++ks) {
          bf16x8 afrag[4], bfrag[4];
#pragma unroll
          for (int i = 0; i < 4; ++i) {
            const int ra = wm + i * 16 + frow;
            const int rb = wn + i * 16 + frow;
            const int cb = ks * 64 + fkb;
            afrag[i] = *reinterpret_cast<const bf16x8*>(
                reinterpret_cast<const char*>(As_) + ra * 128
                + swz(ra, cb));
            bfrag[i] = *reinterpret_cast<const bf16x8*>(
                reinterpret_cast<const char*>(Bs_) + rb * 128
                + swz(rb, cb));
          }
#pragma unroll
          for (int i = 0; i < 4; ++i)
#pragma unroll
            for (int j = 0; j < 4; ++j)
              acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                  afrag[i], bfrag[j], acc[i][j], 0, 0, 0);
        }
      }
#undef GLDS_TILE
    } else {
      // ---- tail tiles: guarded register staging into buffer 0, linear
      // layout with the same swizzled addressing (write side applies
      // the swizzle so the read side is uniform) ----
      for (int kt = 0; kt < KT; ++kt) {
        __syncthreads();
#pragma unroll
        for (int s = 0; s < 2; ++s) {
          const int seg = tid + s * 256;   // 512 segs x 16 B = A+B tile
          const int half = seg >> 8;       // 0: A, 1: B
          const int idx = seg & 255;       // 128 rows x 2 halves... rows*2
          const int row = idx >> 1;
          const int cb = (idx & 1) * 64 + 0;  // two 64-B pieces per row
          // stage 64 B per segment as 4 x 16 B guarded pieces
#pragma unroll
          for (int p = 0; p < 4; ++p) {
            const int cbb = cb + p * 16;
            ushort_t tmp[8];
            const int64_t gr = (half ? (int64_t)tn : tm) + row;
            const int64_t lim = half ? (int64_t)N : M;
            const ushort_t* base = half ? B : A;
#pragma unroll
            for (int j = 0; j < 8; ++j) {
              tmp[j] = (gr < lim)
                  ? base[gr * K + (int64_t)kt * BK3 + cbb / 2 + j]
                  : (ushort_t)0;
            }
            *reinterpret_cast<bf16x8*>(
                reinterpret_cast<char*>(lds) + half * BM * BK3 * 2
                + row * 128 + swz(row, cbb)) =
                *reinterpret_cast<bf16x8*>(tmp);
          }
        }
        __syncthreads();
        const ushort_t* As_ = lds;
        const ushort_t* Bs_ = lds + BM * BK3;
#pragma unroll
        for (int ks = 0; ks < 2; ++ks) {
          bf16x8 afrag[4], bfrag[4];
#pragma unroll
          for (int i = 0; i < 4; ++i) {
            const int ra = wm + i * 16 + frow;
            const int rb = wn + i * 16 + frow;
            const int cb = ks * 64 + fkb;
            afrag[i] = *reinterpret_cast<const bf16x8*>(
                reinterpret_cast<const char*>(As_) + ra * 128
                + swz(ra, cb));
            bfrag[i] = *reinterpret_cast<const bf16x8*>(
                reinterpret_cast<const char*>(Bs_) + rb * 128
                + swz(rb, cb));
          }
#pragma unroll
          for (int i = 0; i < 4; ++i)
#pragma unroll
            for (int j = 0; j < 4; ++j)
              acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                  afrag[i], bfrag[j], acc[i][j], 0, 0, 0);
        }
      }
    }

#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int64_t gr = tm + wm + i * 16 + (lane >> 4) * 4 + r;
          const int gc = tn + wn + j * 16 + (lane & 15);
          if (gr < M && gc < N) C[gr * N + gc] = f2b(acc[i][j][r]);
        }
  }
}

// ----------------------------------- gemm v4 (3-buffer glds, raw barrier)
// Full-tile-only variant: keeps one tile of glds in flight ACROSS the
// barrier (guide: __syncthreads() drains glds via its vmcnt(0) fence —
// the ~20% stall of the 2-buffer structure; raw s_barrier + counted
// s_waitcnt vmcnt(N) leaves the prefetch in flight).  Each wave issues
// 8 glds per tile; vmcnt(8) at the top of the loop means "my tile-k
// chunks have landed, tile-k+1's 8 are still flying"; every wave waits
// its own counter before the barrier, so after the barrier all waves'
// tile-k chunks are visible.  Requires M%128==0, N%128==0, K%64==0,
// K/64 >= 2 (caller-checked).
__global__ __launch_bounds__(256) void k_gemm_nt_bf16_v4(
    const ushort_t* __restrict__ A, const ushort_t* __restrict__ B,
    ushort_t* __restrict__ C, int64_t M, int N, int K) {
  __shared__ ushort_t lds[3 * 2 * BM * BK3];  // 3 buffers x (A,B)

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int wm = (wave >> 1) * 64;
  const int wn = (wave & 1) * 64;
  const int frow = lane & 15;
  const int fkb = (lane >> 4) * 16;
  const int g_row_in_chunk = lane >> 3;
  const int g_cb = (lane & 7) * 16;

  const int n_tiles = N / BN;
  const int64_t m_tiles = M / BM;
  const int64_t total_tiles = m_tiles * n_tiles;
  const int KT = K / BK3;

#define GLDS4(buf, k0)                                                     \
  do {                                                                     \
    _Pragma("unroll") for (int c = 0; c < 4; ++c) {                        \
      const int row = (wave * 4 + c) * 8 + g_row_in_chunk;                 \
      const int scb = swz(row, g_cb);                                      \
      __builtin_amdgcn_global_load_lds(                                    \
          (const __attribute__((address_space(1))) unsigned int*)(         \
              A + (tm + row) * K + (k0) + scb / 2),                        \
          (__attribute__((address_space(3))) unsigned int*)(               \
              lds + (buf) * 2 * BM * BK3 + (wave * 4 + c) * 512),          \
          16, 0, 0);                                                       \
      __builtin_amdgcn_global_load_lds(                                    \
          (const __attribute__((address_space(1))) unsigned int*)(         \
              B + (int64_t)(tn + row) * K + (k0) + scb / 2),               \
          (__attribute__((address_space(3))) unsigned int*)(               \
              lds + ((buf) * 2 + 1) * BM * BK3 + (wave * 4 + c) * 512),    \
          16, 0, 0);                                                       \
    }                                                                      \
  } while (0)

  for (int64_t tile = blockIdx.x; tile < total_tiles; tile += gridDim.x) {
    const int64_t tm = (tile / n_tiles) * BM;
    const int tn = (int)(tile % n_tiles) * BN;

    f32x4 acc[4][4];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j) acc[i][j] = (f32x4){0.f, 0.f, 0.f, 0.f};

    GLDS4(0, 0);
    if (KT > 1) GLDS4(1, BK3);

    for (int kt = 0; kt < KT; ++kt) {
      // wait for OWN tile-k chunks; leave tile-k+1's (if any) in flight
      if (kt + 1 < KT)
        asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
      else
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_barrier();

      const int buf = kt % 3;
      const ushort_t* As_ = lds + buf * 2 * BM * BK3;
      const ushort_t* Bs_ = lds + (buf * 2 + 1) * BM * BK3;
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        bf16x8 afrag[4], bfrag[4];
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          const int ra = wm + i * 16 + frow;
          const int rb = wn + i * 16 + frow;
          const int cb = ks * 64 + fkb;
          afrag[i] = *reinterpret_cast<const bf16x8*>(
              reinterpret_cast<const char*>(As_) + ra * 128 + swz(ra, cb));
          bfrag[i] = *reinterpret_cast<const bf16x8*>(
              reinterpret_cast<const char*>(Bs_) + rb * 128 + swz(rb, cb));
        }
#pragma unroll
        for (int i = 0; i < 4; ++i)
#pragma unroll
          for (int j = 0; j < 4; ++j)
            acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                afrag[i], bfrag[j], acc[i][j], 0, 0, 0);
      }
      // re-fill buffer (kt+2)%3 == (kt-1)%3: passing this iteration's
      // barrier proves every wave finished reading tile k-1, so no
      // second barrier is needed before overwriting its buffer
      if (kt + 2 < KT) GLDS4((kt + 2) % 3, (int64_t)(kt + 2) * BK3);
    }

#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int64_t gr = tm + wm + i * 16 + (lane >> 4) * 4 + r;
          const int gc = tn + wn + j * 16 + (lane & 15);
          C[gr * N + gc] = f2b(acc[i][j][r]);
        }
  }
#undef GLDS4
}

// ----------------------------------------------- gemm NT split-K (v2s)
// v2's register-staged 128x128 geometry with the K reduction split over
// `split` block groups writing fp32 partials [split][M][N] (reduced to
// bf16 by k_gemm_nt_reduce).  For under-filled shapes (the 7x7-stage
// 1x1 convs: M = 1568 -> 13 m-tiles) where neither glds pipeline can
// fill the chip.
__global__ __launch_bounds__(256) void k_gemm_nt_splitk_bf16(
    const ushort_t* __restrict__ A, const ushort_t* __restrict__ B,
    float* __restrict__ P, int64_t M, int N, int K, int split) {
  constexpr int BKT = 64;
  constexpr int STRIDE = BKT + 8;
  constexpr int SEGS = BM * BKT / 8 / 256;
  __shared__ ushort_t As[2][BM * STRIDE];
  __shared__ ushort_t Bs[2][BN * STRIDE];

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int wm = (wave >> 1) * 64;
  const int wn = (wave & 1) * 64;
  const int frow = lane & 15;
  const int fk0 = (lane >> 4) * 8;

  constexpr int SEG_PER_ROW = BKT / 8;
  int rr[SEGS], cc[SEGS];
#pragma unroll
  for (int s = 0; s < SEGS; ++s) {
    const int seg = tid + s * 256;
    rr[s] = seg / SEG_PER_ROW;
    cc[s] = (seg % SEG_PER_ROW) * 8;
  }

  const int n_tiles = (N + BN - 1) / BN;
  const int64_t m_tiles = (M + BM - 1) / BM;
  const int64_t total = m_tiles * n_tiles * split;
  const int KT_all = K / BKT;

  const int64_t bid0 = xcd_virtual_bid();
  for (int64_t t = bid0; t < total; t += gridDim.x) {
    const int ks = (int)(t % split);
    const int64_t ct = t / split;
    const int64_t tm = (ct / n_tiles) * BM;
    const int tn = (int)(ct % n_tiles) * BN;
    const int kt_lo = (int)(((int64_t)KT_all * ks) / split);
    const int kt_hi = (int)(((int64_t)KT_all * (ks + 1)) / split);
    const int KT = kt_hi - kt_lo;
    if (KT <= 0) continue;

    f32x4 acc[4][4];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j) acc[i][j] = (f32x4){0.f, 0.f, 0.f, 0.f};

    bf16x8 ra[SEGS], rb[SEGS];
    const bool full = (tm + BM <= M) && (tn + BN <= N);

#define LOAD_T(k0)                                                         \
  do {                                                                     \
    if (full) {                                                            \
      _Pragma("unroll") for (int s = 0; s < SEGS; ++s) {                   \
        ra[s] = *reinterpret_cast<const bf16x8*>(                          \
            A + (tm + rr[s]) * K + (k0) + cc[s]);                          \
        rb[s] = *reinterpret_cast<const bf16x8*>(                          \
            B + (int64_t)(tn + rr[s]) * K + (k0) + cc[s]);                 \
      }                                                                    \
    } else {                                                               \
      _Pragma("unroll") for (int s = 0; s < SEGS; ++s) {                   \
        const int64_t ga = tm + rr[s];                                     \
        const int gb = tn + rr[s];                                         \
        ushort_t ta[8], tb[8];                                             \
        _Pragma("unroll") for (int j = 0; j < 8; ++j) {                    \
          ta[j] = (ga < M) ? A[ga * K + (k0) + cc[s] + j] : (ushort_t)0;   \
          tb[j] = (gb < N) ? B[(int64_t)gb * K + (k0) + cc[s] + j]         \
                           : (ushort_t)0;                                  \
        }                                                                  \
        ra[s] = *reinterpret_cast<bf16x8*>(ta);                            \
        rb[s] = *reinterpret_cast<bf16x8*>(tb);                            \
      }                                                                    \
    }                                                                      \
  } while (0)

#define WRITE_T(buf)                                                       \
  do {                                                                     \
    _Pragma("unroll") for (int s = 0; s < SEGS; ++s) {                     \
      *reinterpret_cast<bf16x8*>(As[buf] + rr[s] * STRIDE + cc[s]) =       \
          ra[s];                                                           \
      *reinterpret_cast<bf16x8*>(Bs[buf] + rr[s] * STRIDE + cc[s]) =       \
          rb[s];                                                           \
    }                                                                      \
  } while (0)

    LOAD_T((int64_t)kt_lo * BKT);
    WRITE_T(0);

    for (int kt = 0; kt < KT; ++kt) {
      __syncthreads();
      const int buf = kt & 1;
      if (kt + 1 < KT) LOAD_T((int64_t)(kt_lo + kt + 1) * BKT);

#pragma unroll
      for (int kk = 0; kk < BKT / 32; ++kk) {
        bf16x8 afrag[4], bfrag[4];
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          afrag[i] = *reinterpret_cast<const bf16x8*>(
              As[buf] + (wm + i * 16 + frow) * STRIDE + kk * 32 + fk0);
          bfrag[i] = *reinterpret_cast<const bf16x8*>(
              Bs[buf] + (wn + i * 16 + frow) * STRIDE + kk * 32 + fk0);
        }
#pragma unroll
        for (int i = 0; i < 4; ++i)
#pragma unroll
          for (int j = 0; j < 4; ++j)
            acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                afrag[i], bfrag[j], acc[i][j], 0, 0, 0);
      }

      __syncthreads();
      if (kt + 1 < KT) WRITE_T(buf ^ 1);
    }
#undef LOAD_T
#undef WRITE_T

#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int64_t gr = tm + wm + i * 16 + (lane >> 4) * 4 + r;
          const int gc = tn + wn + j * 16 + (lane & 15);
          if (gr < M && gc < N)
            P[((int64_t)ks * M + gr) * N + gc] = acc[i][j][r];
        }
  }
}

__global__ void k_gemm_nt_reduce_bf16(const float* __restrict__ P,
                                      ushort_t* __restrict__ C,
                                      int64_t numel, int split) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < numel; i += stride) {
    float acc = 0.f;
    for (int k = 0; k < split; ++k) acc += P[(int64_t)k * numel + i];
    C[i] = f2b(acc);
  }
}

// ------------------------------------- gemm v6 (256x128 tile, 8 waves)
// The 128x128 tile's LDS-staging traffic (2 x 16 KB per 64-K-step for
// 4.2 MFLOP -> 66 flops/byte) caps it below hipBLASLt.  256x128 with
// 512 threads stages 48 KB per step for 2x the flops (87 flops/byte),
// keeps the same per-wave 64x64 accumulator geometry (8 waves in 4x2),
// and stays at 2 waves/SIMD.  One workgroup per CU.
//   SPAN=false: 2 LDS buffers (96 KB), __syncthreads pipeline (drains
//               glds via its vmcnt fence).
//   SPAN=true:  3 buffers (144 KB of the 160 KB LDS), raw s_barrier +
//               counted vmcnt keeps one tile of glds in flight across
//               the barrier — the guide's 1-block/CU-regime lever (v4
//               showed it loses at 2 blocks/CU; here it can win).
// Full tiles only: caller routes M%256 || N%128 || K%64 != 0 to v5.
#define BM6 256
#define BN6 128
#define TILE6 (BM6 + BN6)          // 384 rows of 128 B = 48 KB / buffer

template <bool SPAN, bool LDSEPI>
__global__ __launch_bounds__(512) void k_gemm_nt_bf16_v6(
    const ushort_t* __restrict__ A, const ushort_t* __restrict__ B,
    ushort_t* __restrict__ C, int64_t M, int N, int K) {
  __shared__ ushort_t lds[(SPAN ? 3 : 2) * TILE6 * BK3];

  const int tid = threadIdx.x;
  const int wave = tid >> 6;         // 8 waves in 4x2
  const int lane = tid & 63;
  const int wm = (wave >> 1) * 64;   // 0,64,128,192
  const int wn = (wave & 1) * 64;    // 0,64
  const int frow = lane & 15;
  const int fkb = (lane >> 4) * 16;
  const int g_row_in_chunk = lane >> 3;
  const int g_cb = (lane & 7) * 16;

  const int n_tiles = N / BN6;
  const int64_t m_tiles = M / BM6;
  const int64_t total_tiles = m_tiles * n_tiles;
  const int KT = K / BK3;

  // per-wave glds assignment: wave w stages A chunks 4w..4w+3 and
  // B chunks 2w..2w+1 (chunk = 8 rows x 128 B = 1 KB); 6 glds per wave
  // per tile step.
#define GLDS6(buf, k0)                                                     \
  do {                                                                     \
    _Pragma("unroll") for (int c = 0; c < 4; ++c) {                        \
      const int row = (wave * 4 + c) * 8 + g_row_in_chunk;                 \
      const int scb = swz(row, g_cb);                                      \
      __builtin_amdgcn_global_load_lds(                                    \
          (const __attribute__((address_space(1))) unsigned int*)(         \
              A + (tm + row) * K + (k0) + scb / 2),                        \
          (__attribute__((address_space(3))) unsigned int*)(               \
              lds + (buf) * TILE6 * BK3 + (wave * 4 + c) * 512),           \
          16, 0, 0);                                                       \
    }                                                                      \
    _Pragma("unroll") for (int c = 0; c < 2; ++c) {                        \
      const int row = (wave * 2 + c) * 8 + g_row_in_chunk;                 \
      const int scb = swz(row, g_cb);                                      \
      __builtin_amdgcn_global_load_lds(                                    \
          (const __attribute__((address_space(1))) unsigned int*)(         \
              B + (int64_t)(tn + row) * K + (k0) + scb / 2),               \
          (__attribute__((address_space(3))) unsigned int*)(               \
              lds + (buf) * TILE6 * BK3 + BM6 * BK3                        \
              + (wave * 2 + c) * 512),                                     \
          16, 0, 0);                                                       \
    }                                                                      \
  } while (0)

  const int64_t bid0 = xcd_virtual_bid();
  for (int64_t tile = bid0; tile < total_tiles; tile += gridDim.x) {
    const int64_t tm = (tile / n_tiles) * BM6;
    const int tn = (int)(tile % n_tiles) * BN6;

    f32x4 acc[4][4];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j) acc[i][j] = (f32x4){0.f, 0.f, 0.f, 0.f};

    if (SPAN) {
      GLDS6(0, 0);
      if (KT > 1) GLDS6(1, BK3);
    } else {
      GLDS6(0, 0);
    }

    for (int kt = 0; kt < KT; ++kt) {
      if (SPAN) {
        if (kt + 1 < KT)
          asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
        else
          asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();
      } else {
        __syncthreads();  // drains in-flight glds (vmcnt fence)
        if (kt + 1 < KT) GLDS6((kt & 1) ^ 1, (int64_t)(kt + 1) * BK3);
      }

      const int buf = SPAN ? (kt % 3) : (kt & 1);
      const ushort_t* As_ = lds + buf * TILE6 * BK3;
      const ushort_t* Bs_ = As_ + BM6 * BK3;
      __builtin_amdgcn_s_setprio(1);  // prioritize the MFMA cluster
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        bf16x8 afrag[4], bfrag[4];
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          const int ra = wm + i * 16 + frow;
          const int rb = wn + i * 16 + frow;
          const int cb = ks * 64 + fkb;
          afrag[i] = *reinterpret_cast<const bf16x8*>(
              reinterpret_cast<const char*>(As_) + ra * 128 + swz(ra, cb));
          bfrag[i] = *reinterpret_cast<const bf16x8*>(
              reinterpret_cast<const char*>(Bs_) + rb * 128 + swz(rb, cb));
        }
#pragma unroll
        for (int i = 0; i < 4; ++i)
#pragma unroll
          for (int j = 0; j < 4; ++j)
            acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                afrag[i], bfrag[j], acc[i][j], 0, 0, 0);
      }
      __builtin_amdgcn_s_setprio(0);
      if (SPAN) {
        // passing this round's barrier proves all waves read tile kt-1:
        // its buffer (kt+2)%3 is safe to refill with no extra barrier
        if (kt + 2 < KT) GLDS6((kt + 2) % 3, (int64_t)(kt + 2) * BK3);
      }
      // non-SPAN: the next iteration's __syncthreads() orders the
      // buffer swap (v3 structure — no trailing barrier needed)
    }
#undef GLDS6

    if (!LDSEPI) {
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const int64_t gr = tm + wm + i * 16 + (lane >> 4) * 4 + r;
            const int gc = tn + wn + j * 16 + (lane & 15);
            C[gr * N + gc] = f2b(acc[i][j][r]);
          }
    } else {
      // LDS-staged epilogue: the tile buffers are dead after the last
      // k-step, so stage the 256x128 bf16 C tile in LDS (scalar
      // conflict-light writes) and stream it out with coalesced 16-B
      // stores (the naive epilogue is 64 scalar 2-B stores per lane).
      ushort_t* cs = lds;  // 256*128*2 B = 64 KB of the freed buffers
      __syncthreads();     // everyone done reading the k-tiles
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const int lr = wm + i * 16 + (lane >> 4) * 4 + r;
            const int lc = wn + j * 16 + (lane & 15);
            cs[lr * BN6 + lc] = f2b(acc[i][j][r]);
          }
      __syncthreads();
      // 256 rows x 256 B; 512 threads -> each thread 4 x 16-B pieces
      const int t = threadIdx.x;
#pragma unroll
      for (int piece = 0; piece < 4; ++piece) {
        const int idx = t + piece * 512;      // 0..2047 16-B segments
        const int row = idx >> 3;             // 8 segments per row
        const int c8 = (idx & 7) * 16;        // bf16 col offset x16
        *reinterpret_cast<bf16x8*>(C + (tm + row) * N + tn + c8) =
            *reinterpret_cast<const bf16x8*>(cs + row * BN6 + c8);
        *reinterpret_cast<bf16x8*>(C + (tm + row) * N + tn + c8 + 8) =
            *reinterpret_cast<const bf16x8*>(cs + row * BN6 + c8 + 8);
      }
      __syncthreads();  // cs is reused as the k-tile image next tile
    }
  }
}

// --------------------------------------------------------- wgrad TN
// dW[Co,Ci] = sum_m dy[m,co] * x[m,ci] — the 1x1-conv weight gradient.
// K = M is huge, so blocks split the M range and store fp32 partials
// [split][128][128]; a second kernel reduces them (deterministic, no
// atomics — same pattern as the BN reductions).  Staging transposes on
// write into the v1 kernel's padded [c][m] LDS image, so the MFMA inner
// loop is identical to the NT kernel's.
//
// Staging geometry (round-2 rewrite after the first hardware numbers
// came in 3-6x below roofline): lanes are M-FASTEST — 16-lane groups
// cover 16 consecutive m-rows of one 8-channel column group, so the
// global read is ONE bf16x8 vector load per lane (rows Co*2 B apart,
// no per-element guards) and the 8 transposed LDS writes land on 8
// consecutive dwords per 16-lane group (2-way conflict) instead of the
// old c-fastest mapping's single bank (16-way).
__global__ __launch_bounds__(256) void k_gemm_tn_partial_bf16(
    const ushort_t* __restrict__ dy,  // [M, Co]
    const ushort_t* __restrict__ x,   // [M, Ci]
    float* __restrict__ partials,     // [split][Co, Ci] (tile-major ok)
    int64_t M, int Co, int Ci, int split) {
  __shared__ ushort_t As[2][BM * LDS_STRIDE];  // [co][m] images (x2)
  __shared__ ushort_t Bs[2][BN * LDS_STRIDE];  // [ci][m]

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int wm = (wave >> 1) * 64;  // co offset of the wave
  const int wn = (wave & 1) * 64;   // ci offset
  const int frow = lane & 15;
  const int fk0 = (lane >> 4) * 8;

  const int co_tiles = (Co + BM - 1) / BM;
  const int ci_tiles = (Ci + BN - 1) / BN;
  const int64_t tiles = (int64_t)co_tiles * ci_tiles * split;

  const int64_t bid0 = xcd_virtual_bid();
  for (int64_t t = bid0; t < tiles; t += gridDim.x) {
    const int s = (int)(t % split);
    const int64_t ct = t / split;
    const int tco = (int)(ct / ci_tiles) * BM;
    const int tci = (int)(ct % ci_tiles) * BN;

    const int64_t m0 = (M * s) / split;
    const int64_t m1 = (M * (s + 1)) / split;

    f32x4 acc[4][4];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j) acc[i][j] = (f32x4){0.f, 0.f, 0.f, 0.f};

    // seg -> (m-row, channel-group), m-fastest within 16-lane groups
    const int mrow = tid & 31;         // 32 m-rows per BK step
    const int cgrp = tid >> 5;         // 8 channel-groups per 256 thr
    const bool a_full = tco + BM <= Co;
    const bool b_full = tci + BN <= Ci;

    // register-staged double buffering (the NT v2 structure): loads
    // for m-chunk t+1 issue right after the barrier and hide behind
    // the MFMA block; writes land in the other LDS buffer
    bf16x8 ra[2], rb[2];

#define TN_LOAD(k0_)                                                       \
  do {                                                                     \
    const int64_t gm = (k0_) + mrow;                                       \
    const bool mok = gm < m1;                                              \
    _Pragma("unroll") for (int half = 0; half < 2; ++half) {               \
      const int c8 = (cgrp + half * 8) * 8;                                \
      {                                                                    \
        const int gc0 = tco + c8;                                          \
        if (mok && a_full) {                                               \
          ra[half] = *reinterpret_cast<const bf16x8*>(dy + gm * Co + gc0); \
        } else {                                                           \
          ushort_t tmp[8];                                                 \
          _Pragma("unroll") for (int j = 0; j < 8; ++j)                    \
            tmp[j] = (mok && gc0 + j < Co)                                 \
                ? dy[gm * Co + gc0 + j] : (ushort_t)0;                     \
          ra[half] = *reinterpret_cast<bf16x8*>(tmp);                      \
        }                                                                  \
      }                                                                    \
      {                                                                    \
        const int gc0 = tci + c8;                                          \
        if (mok && b_full) {                                               \
          rb[half] = *reinterpret_cast<const bf16x8*>(x + gm * Ci + gc0);  \
        } else {                                                           \
          ushort_t tmp[8];                                                 \
          _Pragma("unroll") for (int j = 0; j < 8; ++j)                    \
            tmp[j] = (mok && gc0 + j < Ci)                                 \
                ? x[gm * Ci + gc0 + j] : (ushort_t)0;                      \
          rb[half] = *reinterpret_cast<bf16x8*>(tmp);                      \
        }                                                                  \
      }                                                                    \
    }                                                                      \
  } while (0)

#define TN_WRITE(buf)                                                      \
  do {                                                                     \
    _Pragma("unroll") for (int half = 0; half < 2; ++half) {               \
      const int c8 = (cgrp + half * 8) * 8;                                \
      const ushort_t* ea = reinterpret_cast<const ushort_t*>(&ra[half]);   \
      const ushort_t* eb = reinterpret_cast<const ushort_t*>(&rb[half]);   \
      _Pragma("unroll") for (int j = 0; j < 8; ++j) {                      \
        As[buf][(c8 + j) * LDS_STRIDE + mrow] = ea[j];                     \
        Bs[buf][(c8 + j) * LDS_STRIDE + mrow] = eb[j];                     \
      }                                                                    \
    }                                                                      \
  } while (0)

    const int KTm = (int)((m1 - m0 + BK - 1) / BK);
    TN_LOAD(m0);
    TN_WRITE(0);
    for (int kt = 0; kt < KTm; ++kt) {
      __syncthreads();
      const int buf = kt & 1;
      if (kt + 1 < KTm) TN_LOAD(m0 + (int64_t)(kt + 1) * BK);

      bf16x8 afrag[4], bfrag[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        afrag[i] = *reinterpret_cast<const bf16x8*>(
            As[buf] + (wm + i * 16 + frow) * LDS_STRIDE + fk0);
        bfrag[i] = *reinterpret_cast<const bf16x8*>(
            Bs[buf] + (wn + i * 16 + frow) * LDS_STRIDE + fk0);
      }
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[i], bfrag[j], acc[i][j], 0, 0, 0);

      __syncthreads();
      if (kt + 1 < KTm) TN_WRITE(buf ^ 1);
    }
#undef TN_LOAD
#undef TN_WRITE

    float* out = partials + (int64_t)s * Co * Ci;
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int gco = tco + wm + i * 16 + (lane >> 4) * 4 + r;
          const int gci = tci + wn + j * 16 + (lane & 15);
          if (gco < Co && gci < Ci)
            out[(int64_t)gco * Ci + gci] = acc[i][j][r];
        }
  }
}

__global__ void k_gemm_tn_reduce(const float* __restrict__ partials,
                                 float* __restrict__ dw, int64_t numel,
                                 int split) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < numel; i += stride) {
    float s = 0.f;
    for (int k = 0; k < split; ++k) s += partials[(int64_t)k * numel + i];
    dw[i] = s;
  }
}

inline int gemm_grid(int64_t M, int N) {
  int64_t tiles = ((M + BM - 1) / BM) * (int64_t)((N + BN - 1) / BN);
  if (tiles > 16384) tiles = 16384;
  if (tiles < 1) tiles = 1;
  return (int)tiles;
}

}  // namespace

extern "C" {

void sgp_mfma_probe(const ushort_t* A, const ushort_t* B, float* C,
                    hipStream_t s) {
  hipLaunchKernelGGL(k_mfma_probe, dim3(1), dim3(64), 0, s, A, B, C);
}

void sgp_gemm_nt_bf16(const ushort_t* A, const ushort_t* B, ushort_t* C,
                      int64_t M, int N, int K, hipStream_t s) {
  hipLaunchKernelGGL(k_gemm_nt_bf16, dim3(gemm_grid(M, N)), dim3(256), 0, s,
                     A, B, C, M, N, K);
}

void sgp_gemm_nt_bf16_v3(const ushort_t* A, const ushort_t* B, ushort_t* C,
                         int64_t M, int N, int K, hipStream_t s) {
  hipLaunchKernelGGL(k_gemm_nt_bf16_v3<false>, dim3(gemm_grid(M, N)),
                     dim3(256), 0, s, A, B, C, M, N, K);
}

void sgp_gemm_nt_bf16_v5(const ushort_t* A, const ushort_t* B, ushort_t* C,
                         int64_t M, int N, int K, hipStream_t s) {
  // grid must be a multiple of 8 for the XCD remap
  int grid = (gemm_grid(M, N) + 7) & ~7;
  hipLaunchKernelGGL(k_gemm_nt_bf16_v3<true>, dim3(grid), dim3(256), 0, s,
                     A, B, C, M, N, K);
}

void sgp_gemm_nt_splitk_bf16(const ushort_t* A, const ushort_t* B,
                             float* P, ushort_t* C, int64_t M, int N,
                             int K, int split, hipStream_t s) {
  const int64_t tiles =
      ((M + BM - 1) / BM) * (int64_t)((N + BN - 1) / BN) * split;
  int grid = (int)(tiles > 16384 ? 16384 : tiles);
  grid = (grid + 7) & ~7;
  if (grid < 8) grid = 8;
  hipLaunchKernelGGL(k_gemm_nt_splitk_bf16, dim3(grid), dim3(256), 0, s,
                     A, B, P, M, N, K, split);
  const int64_t numel = M * N;
  int rgrid = (int)(((numel + 255) / 256) > 8192 ? 8192
                                                 : (numel + 255) / 256);
  hipLaunchKernelGGL(k_gemm_nt_reduce_bf16, dim3(rgrid < 1 ? 1 : rgrid),
                     dim3(256), 0, s, P, C, numel, split);
}

void sgp_gemm_nt_bf16_v6(const ushort_t* A, const ushort_t* B, ushort_t* C,
                         int64_t M, int N, int K, int span, hipStream_t s) {
  int64_t tiles = (M / BM6) * (int64_t)(N / BN6);
  if (tiles > 16384) tiles = 16384;
  int grid = (int)((tiles + 7) & ~7);
  if (grid < 8) grid = 8;
  if (span)
    hipLaunchKernelGGL((k_gemm_nt_bf16_v6<true, false>), dim3(grid),
                       dim3(512), 0, s, A, B, C, M, N, K);
  else
    hipLaunchKernelGGL((k_gemm_nt_bf16_v6<false, false>), dim3(grid),
                       dim3(512), 0, s, A, B, C, M, N, K);
}

void sgp_gemm_nt_bf16_v7(const ushort_t* A, const ushort_t* B, ushort_t* C,
                         int64_t M, int N, int K, int span, hipStream_t s) {
  int64_t tiles = (M / BM6) * (int64_t)(N / BN6);
  if (tiles > 16384) tiles = 16384;
  int grid = (int)((tiles + 7) & ~7);
  if (grid < 8) grid = 8;
  if (span)
    hipLaunchKernelGGL((k_gemm_nt_bf16_v6<true, true>), dim3(grid),
                       dim3(512), 0, s, A, B, C, M, N, K);
  else
    hipLaunchKernelGGL((k_gemm_nt_bf16_v6<false, true>), dim3(grid),
                       dim3(512), 0, s, A, B, C, M, N, K);
}

void sgp_gemm_nt_bf16_v4(const ushort_t* A, const ushort_t* B, ushort_t* C,
                         int64_t M, int N, int K, hipStream_t s) {
  hipLaunchKernelGGL(k_gemm_nt_bf16_v4, dim3(gemm_grid(M, N)), dim3(256), 0,
                     s, A, B, C, M, N, K);
}

void sgp_gemm_tn_wgrad_bf16(const ushort_t* dy, const ushort_t* x,
                            float* partials, float* dw, int64_t M, int Co,
                            int Ci, int split, hipStream_t s) {
  const int64_t tiles =
      (int64_t)((Co + 127) / 128) * ((Ci + 127) / 128) * split;
  int grid = (int)(tiles > 16384 ? 16384 : tiles);
  grid = (grid + 7) & ~7;  // multiple of 8: XCD remap bijectivity
  hipLaunchKernelGGL(k_gemm_tn_partial_bf16, dim3(grid), dim3(256), 0, s,
                     dy, x, partials, M, Co, Ci, split);
  const int64_t numel = (int64_t)Co * Ci;
  int rgrid = (int)(((numel + 255) / 256) > 8192 ? 8192
                                                 : (numel + 255) / 256);
  hipLaunchKernelGGL(k_gemm_tn_reduce, dim3(rgrid < 1 ? 1 : rgrid),
                     dim3(256), 0, s, partials, dw, numel, split);
}

void sgp_gemm_nt_bf16_v2(const ushort_t* A, const ushort_t* B, ushort_t* C,
                         int64_t M, int N, int K, hipStream_t s) {
  if (K % 64 == 0)
    hipLaunchKernelGGL((k_gemm_nt_bf16_v2<64>), dim3(gemm_grid(M, N)),
                       dim3(256), 0, s, A, B, C, M, N, K);
  else
    hipLaunchKernelGGL((k_gemm_nt_bf16_v2<32>), dim3(gemm_grid(M, N)),
                       dim3(256), 0, s, A, B, C, M, N, K);
}

}  // extern "C"
