// dense_kernels.hip — fused MLP linear layers for gfx950 (CDNA4 MFMA).
//
// DLRM-class MLPs are small (N,K ∈ [16, 512], M = batch): the BLAS library
// picks low-occupancy tiles for them (measured 147us for an 8192x512x16
// layer and 54us for K=8192 weight grads at 12% occupancy). These kernels
// are shaped for exactly this regime instead:
//   fwd:  C[M,N]  = act(A[M,K] @ W[N,K]^T + bias)      (torch Linear layout)
//   dX:   dX[M,K] = G[M,N] @ W[N,K]                    (G = grad ⊙ act mask)
//   dW:   dW[N,K] = G^T[N,M] @ A[M,K], split over M chunks with fp32
//         atomic accumulation (plenty of workgroups even for N=K=256),
//         fused column-sum for dbias.
//
// MFMA: v_mfma_f32_16x16x32_bf16 per-wave tiles; fragment layouts per the
// CDNA4 ISA (A: row=lane%16, k=8*(lane/16)+i; B: col=lane%16, same k;
// D: col=lane%16, row=4*(lane/16)+i) — verified on-device against a torch
// fp32 reference (tests/test_gpu_dense.py, asymmetric random inputs).
// Memory strategy per kernel (all measured on the DLRM layer mix):
//   fwd: operands straight from L2 (weights are L2-resident, A rows
//        stream); 64x64 block tiles maximize block count.
//   dX:  W columns are strided -> LDS-staged transposed W tile for the
//        wide-N layers (k_linear_dx_lds), direct for narrow ones.
//   dW:  both operands are column reads -> transposed LDS chunks + multi-
//        chunk register accumulation to bound the fp32 atomic flush
//        volume (k_linear_dw_lds64t); 16-col variant for narrow N.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <cstdint>

namespace py = pybind11;

static inline hipStream_t dense_stream() {
  return at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
}

namespace {

using bf16 = __hip_bfloat16;
using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

__device__ __forceinline__ float bf2f_u16(short u) {
  union {
    unsigned int i;
    float f;
  } cv;
  cv.i = ((unsigned int)(unsigned short)u) << 16;
  return cv.f;
}

__device__ __forceinline__ short f2bf_u16(float f) {
  union {
    float f;
    unsigned int i;
  } cv;
  cv.f = f;
  // round-to-nearest-even
  unsigned int lsb = (cv.i >> 16) & 1;
  cv.i += 0x7fff + lsb;
  return (short)(cv.i >> 16);
}

// Load an 8-wide bf16 fragment row: src row-major [rows, cols], fragment
// element i = src[r][k0 + i]; ZERO-FILLED outside bounds.
__device__ __forceinline__ bf16x8 load_frag_row(const short* __restrict__ src,
                                                int r, int k0, int rows,
                                                int cols) {
  bf16x8 out;
  if (r < rows && k0 + 7 < cols) {
    const short* p = src + (int64_t)r * cols + k0;
    // 16B vector load
    out = *reinterpret_cast<const bf16x8*>(p);
  } else {
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      short v = 0;
      if (r < rows && k0 + i < cols) v = src[(int64_t)r * cols + k0 + i];
      out[i] = v;
    }
  }
  return out;
}

// Strided fragment: element i = src[k0 + i][c] (column c of row-major src).
__device__ __forceinline__ bf16x8 load_frag_col(const short* __restrict__ src,
                                                int c, int k0, int rows,
                                                int cols) {
  bf16x8 out;
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    short v = 0;
    int r = k0 + i;
    if (r < rows && c < cols) v = src[(int64_t)r * cols + c];
    out[i] = v;
  }
  return out;
}

// ------------------------------------------------------------------
// fwd: C[M,N] = act(A[M,K] @ W[N,K]^T + bias); 4 waves/block stacked on M.
// MT = m-subtiles per wave: each wave computes MT*16 rows x 64 cols, so
// one set of 4 B fragments feeds 4*MT MFMAs. MT=1 maximizes block count
// (latency hiding via parallelism); MT=2 halves B-fragment loads per
// MFMA. (A 128-col NT=8 variant measured WORSE — 44us vs 29us on
// 8192x512x480 — fewer blocks cost more than the halved A re-reads.)
// ------------------------------------------------------------------
template <int MT>
__global__ void k_linear_fwd_t(const short* __restrict__ A,
                               const short* __restrict__ W,
                               const float* __restrict__ bias, int M, int N,
                               int K, int act, short* __restrict__ C) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int ntiles64 = (N + 63) / 64;
  const int m0 = (blockIdx.x / ntiles64) * 64 * MT + wave * 16 * MT;
  const int n0 = (blockIdx.x % ntiles64) * 64;
  if (m0 >= M) return;
  const int col_b = n0 + (lane & 15);
  const int kgrp = (lane >> 4) * 8;
  f32x4 acc[MT][4] = {};
  const int nt = min(4, (N - n0 + 15) / 16);
  for (int k0 = 0; k0 < K; k0 += 32) {
    bf16x8 a[MT];
#pragma unroll
    for (int r = 0; r < MT; ++r)
      a[r] = load_frag_row(A, m0 + r * 16 + (lane & 15), k0 + kgrp, M, K);
#pragma unroll
    for (int t = 0; t < 4; ++t) {
      if (t >= nt) break;
      bf16x8 b = load_frag_row(W, col_b + t * 16, k0 + kgrp, N, K);
#pragma unroll
      for (int r = 0; r < MT; ++r)
        acc[r][t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[r], b,
                                                            acc[r][t], 0, 0,
                                                            0);
    }
  }
  // D: col = lane%16, row = 4*(lane/16) + i. The MFMA output layout
  // writes 32B fragments per instruction; for FULL 16x64 tiles, stage
  // the tile in LDS and flush 128B row bursts (the fragmented stores
  // capped this kernel at ~520 GB/s on an 8 MB output).
  __shared__ short cstage[4][16][72];  // 72 = 64 + 8 (bank de-phase)
  const bool full_tile =
      (MT == 1) && (n0 + 64 <= N) && (m0 + 16 <= M) && ((N & 7) == 0);
  if (full_tile) {
#pragma unroll
    for (int t = 0; t < 4; ++t) {
      const int cn = t * 16 + (lane & 15);
      const float bv = bias ? bias[n0 + cn] : 0.0f;
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        float v = acc[0][t][i] + bv;
        if (act == 1 && v < 0.0f) v = 0.0f;
        else if (act == 2) v = 1.0f / (1.0f + __expf(-v));
        cstage[wave][(lane >> 4) * 4 + i][cn] = f2bf_u16(v);
      }
    }
    __builtin_amdgcn_s_waitcnt(0);  // LDS writes visible within the wave
    // flush: 8 lanes per row x 8 cols each = one 16B store per lane,
    // two row-groups of 8 -> full 128B bursts per row
#pragma unroll
    for (int half = 0; half < 2; ++half) {
      const int row = half * 8 + (lane >> 3);
      const int c8 = (lane & 7) * 8;
      bf16x8 vreg;
#pragma unroll
      for (int j = 0; j < 8; ++j) vreg[j] = cstage[wave][row][c8 + j];
      *reinterpret_cast<bf16x8*>(
          C + (int64_t)(m0 + row) * N + n0 + c8) = vreg;
    }
    return;
  }
#pragma unroll
  for (int r = 0; r < MT; ++r) {
#pragma unroll
    for (int t = 0; t < 4; ++t) {
      if (t >= nt) break;
      const int cn = n0 + t * 16 + (lane & 15);
      if (cn >= N) continue;
      const float bv = bias ? bias[cn] : 0.0f;
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        int cm = m0 + r * 16 + (lane >> 4) * 4 + i;
        if (cm >= M) continue;
        float v = acc[r][t][i] + bv;
        if (act == 1 && v < 0.0f) v = 0.0f;          // relu
        else if (act == 2) v = 1.0f / (1.0f + __expf(-v));  // sigmoid
        C[(int64_t)cm * N + cn] = f2bf_u16(v);
      }
    }
  }
}

// ------------------------------------------------------------------
// dX[M,K] = G[M,N] @ W[N,K]   (GEMM-K = N; B fragment is strided)
// ------------------------------------------------------------------
__global__ void k_linear_dx(const short* __restrict__ G,
                            const short* __restrict__ W, int M, int N, int K,
                            short* __restrict__ dX) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int ktiles64 = (K + 63) / 64;
  const int m0 = (blockIdx.x / ktiles64) * 64 + wave * 16;
  const int c0 = (blockIdx.x % ktiles64) * 64;
  if (m0 >= M) return;
  const int row_g = m0 + (lane & 15);
  const int col_w = c0 + (lane & 15);
  const int kgrp = (lane >> 4) * 8;
  f32x4 acc[4] = {{0.f, 0.f, 0.f, 0.f}, {0.f, 0.f, 0.f, 0.f},
                  {0.f, 0.f, 0.f, 0.f}, {0.f, 0.f, 0.f, 0.f}};
  const int kt = min(4, (K - c0 + 15) / 16);
  for (int n0 = 0; n0 < N; n0 += 32) {
    bf16x8 a = load_frag_row(G, row_g, n0 + kgrp, M, N);
#pragma unroll
    for (int t = 0; t < 4; ++t) {
      if (t >= kt) break;
      bf16x8 b = load_frag_col(W, col_w + t * 16, n0 + kgrp, N, K);
      acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[t], 0, 0,
                                                       0);
    }
  }
#pragma unroll
  for (int t = 0; t < 4; ++t) {
    if (t >= kt) break;
    const int ck = c0 + t * 16 + (lane & 15);
    if (ck >= K) continue;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int cm = m0 + (lane >> 4) * 4 + i;
      if (cm >= M) continue;
      dX[(int64_t)cm * K + ck] = f2bf_u16(acc[t][i]);
    }
  }
}

// ------------------------------------------------------------------
// dX via LDS-staged weights: the 64-col variant above builds its B
// fragments from 8 strided global loads each (W columns); for wide N
// that dominates (measured 37.5us on the 8192x512x480 layer). Here
// W[:, c0:c0+64] is staged TRANSPOSED into LDS once per 256-row N
// superchunk (row stride 264 shorts keeps the 16B fragment reads
// aligned), and each wave computes 32 rows x 64 cols (8 MFMAs per pair
// of G fragments).
// ------------------------------------------------------------------
constexpr int DX_NSUP = 256;  // N rows staged per superchunk (33 KB LDS)
__global__ void k_linear_dx_lds(const short* __restrict__ G,
                                const short* __restrict__ W, int M, int N,
                                int K, short* __restrict__ dX) {
  __shared__ short wl[64][DX_NSUP + 8];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int ktiles64 = (K + 63) / 64;
  const int m0 = (blockIdx.x / ktiles64) * 128 + wave * 32;
  const int c0 = (blockIdx.x % ktiles64) * 64;
  const int row_g0 = m0 + (lane & 15);
  const int row_g1 = m0 + 16 + (lane & 15);
  const int kgrp = (lane >> 4) * 8;
  const int kt = min(4, (K - c0 + 15) / 16);
  f32x4 acc[2][4] = {};
  for (int ns = 0; ns < N; ns += DX_NSUP) {
    const int nlen = min(DX_NSUP, N - ns);
    __syncthreads();  // previous superchunk's reads must finish
    {
      // stage W[ns..ns+nlen][c0..c0+64] -> wl[c][n]; rows past N are
      // zero-filled by load_frag_row so fragment tails read zeros
      const int r = threadIdx.x >> 3;
      const int seg = threadIdx.x & 7;
      const int nstg = (nlen + 31) & ~31;
      for (int it = 0; it * 32 < nstg; ++it) {
        int n = it * 32 + r;
        bf16x8 v = load_frag_row(W, ns + n, c0 + seg * 8, N, K);
#pragma unroll
        for (int j = 0; j < 8; ++j) wl[seg * 8 + j][n] = v[j];
      }
    }
    __syncthreads();
    if (m0 >= M) continue;
    for (int n0 = 0; n0 < nlen; n0 += 32) {
      bf16x8 a0 = load_frag_row(G, row_g0, ns + n0 + kgrp, M, N);
      bf16x8 a1 = load_frag_row(G, row_g1, ns + n0 + kgrp, M, N);
#pragma unroll
      for (int t = 0; t < 4; ++t) {
        if (t >= kt) break;
        bf16x8 b = *reinterpret_cast<const bf16x8*>(
            &wl[t * 16 + (lane & 15)][n0 + kgrp]);
        acc[0][t] =
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b, acc[0][t], 0, 0,
                                                    0);
        acc[1][t] =
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b, acc[1][t], 0, 0,
                                                    0);
      }
    }
  }
  if (m0 >= M) return;
#pragma unroll
  for (int rg = 0; rg < 2; ++rg) {
#pragma unroll
    for (int t = 0; t < 4; ++t) {
      if (t >= kt) break;
      const int ck = c0 + t * 16 + (lane & 15);
      if (ck >= K) continue;
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        int cm = m0 + rg * 16 + (lane >> 4) * 4 + i;
        if (cm >= M) continue;
        dX[(int64_t)cm * K + ck] = f2bf_u16(acc[rg][t][i]);
      }
    }
  }
}

// LDS-staged dW: block = (16-row N tile) x (64-col K group) x (128-row M
// chunk). G[128,16] and A[128,64] chunks are staged with coalesced row
// loads; the four waves compute the four 16x16 k-subtiles from LDS
// (column reads land in distinct banks: 16 consecutive bf16 = 32 B).
// Replaces the 8-scalar-strided-load fragments (31 us -> target <10 us on
// the 8192x512x256 layer). fp32 atomic accumulate; fused dbias.
__global__ void k_linear_dw_lds(const short* __restrict__ G,
                                const short* __restrict__ A, int M, int N,
                                int K, float* __restrict__ dW,
                                float* __restrict__ dbias) {
  constexpr int CH = 128;      // chunk rows
  constexpr int KG = 64;       // K columns per block
  constexpr int APAD = 8;      // LDS row pad (banks)
  __shared__ short a_l[CH][KG + APAD];
  __shared__ short g_l[CH][16 + APAD];
  __shared__ float bred[16];

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int tid = threadIdx.x;
  const int kgroups = (K + KG - 1) / KG;
  const int tiles_n = (N + 15) / 16;
  const int tile_id = blockIdx.x % (tiles_n * kgroups);
  const int chunk = blockIdx.x / (tiles_n * kgroups);
  const int n0 = (tile_id / kgroups) * 16;
  const int k0 = (tile_id % kgroups) * KG;
  const int mbeg = chunk * CH;

  // stage A[mbeg..mbeg+128][k0..k0+64]: 8 lanes x 16B per row, 32 rows/iter
  {
    const int r_in_iter = tid >> 3;        // 0..31
    const int seg = tid & 7;               // 0..7 (16B segments)
    for (int it = 0; it < CH / 32; ++it) {
      int r = it * 32 + r_in_iter;
      int m = mbeg + r;
      bf16x8 v = load_frag_row(A, m, k0 + seg * 8, M, K);
      *reinterpret_cast<bf16x8*>(&a_l[r][seg * 8]) = v;
    }
    // stage G[mbeg..][n0..n0+16]: 2 lanes x 16B per row, 128 rows/iter
    const int r_g = tid >> 1;              // 0..127
    const int seg_g = tid & 1;
    bf16x8 v = load_frag_row(G, mbeg + r_g, n0 + seg_g * 8, M, N);
    *reinterpret_cast<bf16x8*>(&g_l[r_g][seg_g * 8]) = v;
  }
  __syncthreads();

  // wave w computes dW[n0..n0+16][k0+w*16 .. +16]
  const int kw = wave * 16;
  const int col_a = kw + (lane & 15);
  const int col_g = lane & 15;
  const int kgrp = (lane >> 4) * 8;
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  float bsum = 0.0f;
  const bool do_bias = (dbias != nullptr) && (k0 == 0) && (wave == 0);
#pragma unroll
  for (int ms = 0; ms < CH; ms += 32) {
    bf16x8 afrag, gfrag;
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      int m = ms + kgrp + i;
      gfrag[i] = g_l[m][col_g];
      afrag[i] = a_l[m][col_a];
    }
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(gfrag, afrag, acc, 0, 0,
                                                  0);
    if (do_bias) {
#pragma unroll
      for (int i = 0; i < 8; ++i) bsum += bf2f_u16(gfrag[i]);
    }
  }
  const int ck = k0 + col_a;
  if (ck < K) {
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int cn = n0 + (lane >> 4) * 4 + i;
      if (cn >= N) continue;
      atomicAdd(&dW[(int64_t)cn * K + ck], acc[i]);
    }
  }
  if (do_bias) {
    float v = bsum;
    for (int off = 16; off < 64; off += 16)
      v += __shfl(bsum, (lane & 15) + off);
    if (lane < 16) {
      int cn = n0 + lane;
      if (cn < N) atomicAdd(&dbias[cn], v);
    }
  }
}

// 64-col-G variant: block = 64n x 64k x 128m. The 16-col version above
// restages the A chunk once per 16-row N tile (N=512 -> 32x = 252 MB of
// A traffic on the 8192x512x480 layer); staging G and A as 64-wide tiles
// cuts that to N/64 restages (63 MB) + K/64 G restages. Wave w owns
// output rows n0+16w..+16, all 64 k columns.
__global__ void k_linear_dw_lds64(const short* __restrict__ G,
                                  const short* __restrict__ A, int M, int N,
                                  int K, float* __restrict__ dW,
                                  float* __restrict__ dbias) {
  constexpr int CH = 128;
  __shared__ short a_l[CH][64 + 8];
  __shared__ short g_l[CH][64 + 8];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int tid = threadIdx.x;
  const int kgroups = (K + 63) / 64;
  const int ngroups = (N + 63) / 64;
  const int tile_id = blockIdx.x % (ngroups * kgroups);
  const int chunk = blockIdx.x / (ngroups * kgroups);
  const int n0 = (tile_id / kgroups) * 64;
  const int k0 = (tile_id % kgroups) * 64;
  const int mbeg = chunk * CH;
  {
    const int r = tid >> 3, seg = tid & 7;
    for (int it = 0; it < CH / 32; ++it) {
      int rr = it * 32 + r;
      *reinterpret_cast<bf16x8*>(&a_l[rr][seg * 8]) =
          load_frag_row(A, mbeg + rr, k0 + seg * 8, M, K);
      *reinterpret_cast<bf16x8*>(&g_l[rr][seg * 8]) =
          load_frag_row(G, mbeg + rr, n0 + seg * 8, M, N);
    }
  }
  __syncthreads();
  const int col_g = wave * 16 + (lane & 15);
  const int kgrp = (lane >> 4) * 8;
  f32x4 acc[4] = {};
  float bsum = 0.0f;
  const bool do_bias = (dbias != nullptr) && (k0 == 0);
#pragma unroll
  for (int ms = 0; ms < CH; ms += 32) {
    bf16x8 gfrag;
#pragma unroll
    for (int i = 0; i < 8; ++i) gfrag[i] = g_l[ms + kgrp + i][col_g];
    if (do_bias) {
#pragma unroll
      for (int i = 0; i < 8; ++i) bsum += bf2f_u16(gfrag[i]);
    }
#pragma unroll
    for (int t = 0; t < 4; ++t) {
      bf16x8 afrag;
#pragma unroll
      for (int i = 0; i < 8; ++i)
        afrag[i] = a_l[ms + kgrp + i][t * 16 + (lane & 15)];
      acc[t] =
          __builtin_amdgcn_mfma_f32_16x16x32_bf16(gfrag, afrag, acc[t], 0,
                                                  0, 0);
    }
  }
#pragma unroll
  for (int t = 0; t < 4; ++t) {
    const int ck = k0 + t * 16 + (lane & 15);
    if (ck >= K) continue;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int cn = n0 + wave * 16 + (lane >> 4) * 4 + i;
      if (cn >= N) continue;
      atomicAdd(&dW[(int64_t)cn * K + ck], acc[t][i]);
    }
  }
  if (do_bias) {
    // each wave owns 16 distinct bias columns (n0 + 16w + lane%16):
    // reduce the 4 kgrp groups' partials within the wave, one atomic each
    float v = bsum;
    for (int off = 16; off < 64; off += 16)
      v += __shfl(bsum, (lane & 15) + off);
    const int cn = n0 + wave * 16 + (lane & 15);
    if (lane < 16 && cn < N) atomicAdd(&dbias[cn], v);
  }
}

// Transposed-LDS dW: same 64n x 64k tile as lds64, but (a) A and G
// chunks are stored TRANSPOSED ([col][row], row stride 136 shorts keeps
// 16B alignment) so each MFMA fragment is ONE aligned b128 LDS read, and
// (b) each block accumulates `cpb` consecutive 128-row M chunks in
// registers before its single atomic flush. The per-chunk atomic flush
// was the real bound: 64 chunks x N*K fp32 atomics = 63 MB of atomic
// writes on the 8192x512x480 layer (~31 us of the 60 us).
__global__ void k_linear_dw_lds64t(const short* __restrict__ G,
                                   const short* __restrict__ A, int M, int N,
                                   int K, int cpb,
                                   float* __restrict__ dW,
                                   float* __restrict__ dbias) {
  constexpr int CH = 128;
  __shared__ short a_t[64][CH + 8];
  __shared__ short g_t[64][CH + 8];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int tid = threadIdx.x;
  const int kgroups = (K + 63) / 64;
  const int ngroups = (N + 63) / 64;
  const int tile_id = blockIdx.x % (ngroups * kgroups);
  const int chunk0 = (blockIdx.x / (ngroups * kgroups)) * cpb;
  const int n0 = (tile_id / kgroups) * 64;
  const int k0 = (tile_id % kgroups) * 64;
  const int col_g = wave * 16 + (lane & 15);
  const int kgrp = (lane >> 4) * 8;
  f32x4 acc[4] = {};
  float bsum = 0.0f;
  const bool do_bias = (dbias != nullptr) && (k0 == 0);
  for (int c = 0; c < cpb; ++c) {
    const int mbeg = (chunk0 + c) * CH;
    if (mbeg >= M) break;
    if (c > 0) __syncthreads();  // previous chunk's reads must finish
    {
      // coalesced global row loads, transposing scalar stores into LDS
      const int r = tid >> 3, seg = tid & 7;
      for (int it = 0; it < CH / 32; ++it) {
        int rr = it * 32 + r;
        bf16x8 va = load_frag_row(A, mbeg + rr, k0 + seg * 8, M, K);
        bf16x8 vg = load_frag_row(G, mbeg + rr, n0 + seg * 8, M, N);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          a_t[seg * 8 + j][rr] = va[j];
          g_t[seg * 8 + j][rr] = vg[j];
        }
      }
    }
    __syncthreads();
#pragma unroll
    for (int ms = 0; ms < CH; ms += 32) {
      bf16x8 gfrag =
          *reinterpret_cast<const bf16x8*>(&g_t[col_g][ms + kgrp]);
      if (do_bias) {
#pragma unroll
        for (int i = 0; i < 8; ++i) bsum += bf2f_u16(gfrag[i]);
      }
#pragma unroll
      for (int t = 0; t < 4; ++t) {
        bf16x8 afrag = *reinterpret_cast<const bf16x8*>(
            &a_t[t * 16 + (lane & 15)][ms + kgrp]);
        acc[t] =
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(gfrag, afrag, acc[t],
                                                    0, 0, 0);
      }
    }
  }
#pragma unroll
  for (int t = 0; t < 4; ++t) {
    const int ck = k0 + t * 16 + (lane & 15);
    if (ck >= K) continue;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int cn = n0 + wave * 16 + (lane >> 4) * 4 + i;
      if (cn >= N) continue;
      atomicAdd(&dW[(int64_t)cn * K + ck], acc[t][i]);
    }
  }
  if (do_bias) {
    float v = bsum;
    for (int off = 16; off < 64; off += 16)
      v += __shfl(bsum, (lane & 15) + off);
    const int cn = n0 + wave * 16 + (lane & 15);
    if (lane < 16 && cn < N) atomicAdd(&dbias[cn], v);
  }
}

// ------------------------------------------------------------------
// Fused row L2-normalize: y = x / sqrt(max(sum(x^2), eps)); backward
// dx = (dy - y * <y, dy>) * inv_norm. One wave per row, shfl reduction
// (reference capability: FusedL2Normalize[Grad],
// fused_l2_normalize_op.cc:236,493 — CPU AVX there, CDNA4 wave here).
// ------------------------------------------------------------------
__device__ __forceinline__ float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off; off >>= 1) v += __shfl_down(v, off);
  return __shfl(v, 0);
}

__global__ void k_l2norm_fwd(const short* __restrict__ X, int M, int N,
                             float eps, short* __restrict__ Y,
                             float* __restrict__ inv_norms) {
  const int lane = threadIdx.x & 63;
  const int row = blockIdx.x * 4 + (threadIdx.x >> 6);
  if (row >= M) return;
  const short* x = X + (int64_t)row * N;
  float ss = 0.0f;
  for (int j = lane; j < N; j += 64) {
    float v = bf2f_u16(x[j]);
    ss += v * v;
  }
  float inv = rsqrtf(fmaxf(wave_sum(ss), eps));
  short* y = Y + (int64_t)row * N;
  for (int j = lane; j < N; j += 64)
    y[j] = f2bf_u16(bf2f_u16(x[j]) * inv);
  if (lane == 0) inv_norms[row] = inv;
}

__global__ void k_l2norm_bwd(const short* __restrict__ dY,
                             const short* __restrict__ Y,
                             const float* __restrict__ inv_norms, int M,
                             int N, short* __restrict__ dX) {
  const int lane = threadIdx.x & 63;
  const int row = blockIdx.x * 4 + (threadIdx.x >> 6);
  if (row >= M) return;
  const short* dy = dY + (int64_t)row * N;
  const short* y = Y + (int64_t)row * N;
  float dot = 0.0f;
  for (int j = lane; j < N; j += 64)
    dot += bf2f_u16(y[j]) * bf2f_u16(dy[j]);
  dot = wave_sum(dot);
  const float inv = inv_norms[row];
  short* dx = dX + (int64_t)row * N;
  for (int j = lane; j < N; j += 64)
    dx[j] = f2bf_u16((bf2f_u16(dy[j]) - bf2f_u16(y[j]) * dot) * inv);
}

// zero-fill (see ev_kernels k_zero_f32: memset nodes don't replay in
// captured graphs)
__global__ void k_zero_f32d(float* __restrict__ p, int64_t n) {
  int64_t i = (blockIdx.x * (int64_t)blockDim.x + threadIdx.x) * 4;
  int64_t stride = gridDim.x * (int64_t)blockDim.x * 4;
  for (; i + 3 < n; i += stride)
    *reinterpret_cast<float4*>(p + i) = make_float4(0.f, 0.f, 0.f, 0.f);
  if (blockIdx.x == 0 && threadIdx.x == 0)
    for (int64_t j = n & ~3LL; j < n; ++j) p[j] = 0.0f;
}

// ---------------------------------------------------------------------
// fused flat dense Adam (≙ reference dense ApplyAdamAsync,
// training_ali_ops_gpu.cu.cc:534, re-designed as ONE kernel over a flat
// parameter buffer): bias-correction from device-resident beta powers
// (capture-safe, no elementwise prelude), optional bf16 shadow emission
// in the same pass (replaces the per-step multi-tensor cast), optional
// gradient scale (folds the data-parallel 1/world averaging into the
// update so the all-reduce needs no separate div).
__global__ void k_dense_adam(float* __restrict__ w,
                             const float* __restrict__ g,
                             float* __restrict__ m, float* __restrict__ v,
                             short* __restrict__ w16,
                             const float* __restrict__ powers, int64_t n,
                             float lr, float beta1, float beta2, float eps,
                             float gscale) {
  const float lr_t = lr * sqrtf(1.0f - powers[1]) / (1.0f - powers[0]);
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; i < n; i += stride) {
    float gi = g[i] * gscale;
    float mn = beta1 * m[i] + (1.0f - beta1) * gi;
    float vn = beta2 * v[i] + (1.0f - beta2) * gi * gi;
    m[i] = mn;
    v[i] = vn;
    float wn = w[i] - lr_t * mn / (sqrtf(vn) + eps);
    w[i] = wn;
    if (w16) w16[i] = f2bf_u16(wn);
  }
}

// ---------------------------------------------------------------------
// fused residual + LayerNorm (transformer sublayer epilogue).
// torch's LN kernels cost ~930 us/step on [B*T, 32] rows (BST profile);
// one THREAD per row with N as a template parameter keeps every row
// array in registers (a runtime-bounded local array spills to scratch —
// measured 175/348 us per call; this form is bandwidth-bound).
// fwd also emits z = x + a (bf16) and per-row (mean, rstd) for backward.
template <int N>
__global__ void k_resln_fwd(const short* __restrict__ x,
                            const short* __restrict__ a,
                            const float* __restrict__ gamma,
                            const float* __restrict__ beta, int64_t M,
                            float eps, short* __restrict__ y,
                            short* __restrict__ z,
                            float* __restrict__ stats) {
  float gm[N], bt[N];
#pragma unroll
  for (int j = 0; j < N; ++j) {
    gm[j] = gamma[j];
    bt[j] = beta[j];
  }
  int64_t r = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; r < M; r += stride) {
    const short* xp = x + r * N;
    const short* ap = a + r * N;
    float zv[N];
    float mean = 0.0f;
#pragma unroll
    for (int j = 0; j < N; ++j) {
      zv[j] = bf2f_u16(xp[j]) + bf2f_u16(ap[j]);
      mean += zv[j];
    }
    mean /= N;
    float var = 0.0f;
#pragma unroll
    for (int j = 0; j < N; ++j) {
      float d = zv[j] - mean;
      var += d * d;
    }
    float rstd = rsqrtf(var / N + eps);
    short* yp = y + r * N;
    short* zp = z + r * N;
#pragma unroll
    for (int j = 0; j < N; ++j) {
      zp[j] = f2bf_u16(zv[j]);
      yp[j] = f2bf_u16((zv[j] - mean) * rstd * gm[j] + bt[j]);
    }
    stats[r * 2] = mean;
    stats[r * 2 + 1] = rstd;
  }
}

// backward: dz per row (flows to BOTH residual inputs). dgamma/dbeta
// partials accumulate in REGISTERS across this thread's rows; one LDS
// merge + one global atomic per (block, column) at the end.
template <int N>
__global__ void k_resln_bwd(const short* __restrict__ dy,
                            const short* __restrict__ z,
                            const float* __restrict__ stats,
                            const float* __restrict__ gamma, int64_t M,
                            short* __restrict__ dz,
                            float* __restrict__ dgamma,
                            float* __restrict__ dbeta) {
  __shared__ float ldg[N];
  __shared__ float ldb[N];
  for (int j = threadIdx.x; j < N; j += blockDim.x) {
    ldg[j] = 0.0f;
    ldb[j] = 0.0f;
  }
  __syncthreads();
  float gm[N], pg[N], pb[N];
#pragma unroll
  for (int j = 0; j < N; ++j) {
    gm[j] = gamma[j];
    pg[j] = 0.0f;
    pb[j] = 0.0f;
  }
  int64_t r = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; r < M; r += stride) {
    const short* dyp = dy + r * N;
    const short* zp = z + r * N;
    const float mean = stats[r * 2];
    const float rstd = stats[r * 2 + 1];
    float s1 = 0.0f, s2 = 0.0f;
    float xh[N], g[N];
#pragma unroll
    for (int j = 0; j < N; ++j) {
      xh[j] = (bf2f_u16(zp[j]) - mean) * rstd;
      g[j] = bf2f_u16(dyp[j]);
      float gg = g[j] * gm[j];
      s1 += gg;
      s2 += gg * xh[j];
      pg[j] += g[j] * xh[j];
      pb[j] += g[j];
    }
    s1 /= N;
    s2 /= N;
    short* dzp = dz + r * N;
#pragma unroll
    for (int j = 0; j < N; ++j)
      dzp[j] = f2bf_u16((g[j] * gm[j] - s1 - xh[j] * s2) * rstd);
  }
#pragma unroll
  for (int j = 0; j < N; ++j) {
    atomicAdd(&ldg[j], pg[j]);
    atomicAdd(&ldb[j], pb[j]);
  }
  __syncthreads();
  for (int j = threadIdx.x; j < N; j += blockDim.x) {
    atomicAdd(&dgamma[j], ldg[j]);
    atomicAdd(&dbeta[j], ldb[j]);
  }
}

// activation backward: G = dY * act_grad(out); act 1=relu, 2=sigmoid
__global__ void k_act_bwd(const short* __restrict__ dY,
                          const short* __restrict__ out, int64_t n, int act,
                          short* __restrict__ G) {
  int64_t i = (blockIdx.x * (int64_t)blockDim.x + threadIdx.x) * 8;
  int64_t stride = gridDim.x * (int64_t)blockDim.x * 8;
  for (; i + 7 < n; i += stride) {
    bf16x8 g = *reinterpret_cast<const bf16x8*>(dY + i);
    bf16x8 o = *reinterpret_cast<const bf16x8*>(out + i);
    bf16x8 r;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      if (act == 1) {
        r[j] = bf2f_u16(o[j]) > 0.0f ? g[j] : (short)0;
      } else {
        float ov = bf2f_u16(o[j]);
        r[j] = f2bf_u16(bf2f_u16(g[j]) * ov * (1.0f - ov));
      }
    }
    *reinterpret_cast<bf16x8*>(G + i) = r;
  }
  // tail
  if (blockIdx.x == 0 && threadIdx.x == 0) {
    for (int64_t j = n & ~7LL; j < n; ++j) {
      if (act == 1) {
        G[j] = bf2f_u16(out[j]) > 0.0f ? dY[j] : (short)0;
      } else {
        float ov = bf2f_u16(out[j]);
        G[j] = f2bf_u16(bf2f_u16(dY[j]) * ov * (1.0f - ov));
      }
    }
  }
}

// ------------------------------------------------------------------
// DLRM dot interaction: out[b, p] = <feats[b,i,:], feats[b,j,:]> for
// pairs i<j (upper triangle), padded to P_pad columns with zeros.
// Replaces bmm + triu-gather (library strided-batched GEMM of 27x27x16
// per sample ran at ~325us/step + 63us indexing backward).
// One wave per sample: feats row staged in LDS, each lane computes
// ceil(P/64) pairs. (reference capability: _dot_op, modelzoo/dlrm/
// train.py:121-132 and the Op_dot fusion templates)
// ------------------------------------------------------------------
__global__ void k_interact_fwd(const short* __restrict__ feats, int B, int F,
                               int D, int P, int P_pad,
                               short* __restrict__ out) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int b = blockIdx.x * 4 + wave;
  extern __shared__ short lds[];
  short* f = lds + wave * F * D;
  const bool active = b < B;
  if (active) {
    const short* src = feats + (int64_t)b * F * D;
    for (int t = lane; t * 8 < F * D; t += 64) {
      *reinterpret_cast<bf16x8*>(f + t * 8) =
          *reinterpret_cast<const bf16x8*>(src + t * 8);
    }
  }
  __syncthreads();
  if (!active) return;
  short* dst = reinterpret_cast<short*>(out) + (int64_t)b * P_pad;
  for (int p = lane; p < P_pad; p += 64) {
    if (p >= P) {
      dst[p] = 0;
      continue;
    }
    // pair index -> (i, j), i < j
    int i = 0, rem = p, row = F - 1;
    while (rem >= row) {
      rem -= row;
      --row;
      ++i;
    }
    int j = i + 1 + rem;
    float acc = 0.0f;
    const short* fi = f + i * D;
    const short* fj = f + j * D;
    for (int d = 0; d < D; ++d)
      acc += bf2f_u16(fi[d]) * bf2f_u16(fj[d]);
    dst[p] = f2bf_u16(acc);
  }
}

// backward: dfeats[b,i,d] = sum_j!=i g[b, pair(i,j)] * feats[b,j,d]
// Cat-fused variant: takes bot [B, D] and emb [B, Fe, D] separately and
// emits top_in = [bot | pair dots] [B, D + P_pad] directly — removes the
// two torch cats around the interaction (feats assembly + top-MLP input).
__global__ void k_interact_cat_fwd(const short* __restrict__ bot,
                                   const short* __restrict__ emb, int B,
                                   int Fe, int D, int P, int P_pad,
                                   short* __restrict__ out) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int b = blockIdx.x * 4 + wave;
  const int F = Fe + 1;
  extern __shared__ short lds[];
  short* f = lds + wave * F * D;
  const bool active = b < B;
  if (active) {
    const short* bsrc = bot + (int64_t)b * D;
    for (int t = lane; t * 8 < D; t += 64)
      *reinterpret_cast<bf16x8*>(f + t * 8) =
          *reinterpret_cast<const bf16x8*>(bsrc + t * 8);
    const short* esrc = emb + (int64_t)b * Fe * D;
    for (int t = lane; t * 8 < Fe * D; t += 64)
      *reinterpret_cast<bf16x8*>(f + D + t * 8) =
          *reinterpret_cast<const bf16x8*>(esrc + t * 8);
  }
  __syncthreads();
  if (!active) return;
  short* dst = reinterpret_cast<short*>(out) + (int64_t)b * (D + P_pad);
  for (int d = lane; d < D; d += 64) dst[d] = f[d];  // bot passthrough
  dst += D;
  for (int p = lane; p < P_pad; p += 64) {
    if (p >= P) {
      dst[p] = 0;
      continue;
    }
    int i = 0, rem = p, row = F - 1;
    while (rem >= row) {
      rem -= row;
      --row;
      ++i;
    }
    int j = i + 1 + rem;
    float acc = 0.0f;
    const short* fi = f + i * D;
    const short* fj = f + j * D;
    for (int d = 0; d < D; ++d)
      acc += bf2f_u16(fi[d]) * bf2f_u16(fj[d]);
    dst[p] = f2bf_u16(acc);
  }
}

__global__ void k_interact_cat_bwd(const short* __restrict__ grad,
                                   const short* __restrict__ bot,
                                   const short* __restrict__ emb, int B,
                                   int Fe, int D, int P, int P_pad,
                                   short* __restrict__ dbot,
                                   short* __restrict__ demb) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int b = blockIdx.x * 4 + wave;
  const int F = Fe + 1;
  extern __shared__ short lds[];
  short* f = lds + wave * (F * D + ((P + 7) & ~7));
  short* g = f + F * D;
  const bool active = b < B;
  if (active) {
    const short* bsrc = bot + (int64_t)b * D;
    for (int t = lane; t * 8 < D; t += 64)
      *reinterpret_cast<bf16x8*>(f + t * 8) =
          *reinterpret_cast<const bf16x8*>(bsrc + t * 8);
    const short* esrc = emb + (int64_t)b * Fe * D;
    for (int t = lane; t * 8 < Fe * D; t += 64)
      *reinterpret_cast<bf16x8*>(f + D + t * 8) =
          *reinterpret_cast<const bf16x8*>(esrc + t * 8);
    const short* gsrc = grad + (int64_t)b * (D + P_pad) + D;
    for (int t = lane; t < P; t += 64) g[t] = gsrc[t];
  }
  __shared__ short plut[40 * 40];
  for (int t = threadIdx.x; t < F * F; t += blockDim.x) {
    int i = t / F, j = t % F;
    int lo = i < j ? i : j, hi = i < j ? j : i;
    plut[t] = (short)(i == j ? -1
                             : lo * F - (lo * (lo + 1)) / 2 + (hi - lo - 1));
  }
  __syncthreads();
  if (!active) return;
  const short* gdirect = grad + (int64_t)b * (D + P_pad);
  short* db = dbot + (int64_t)b * D;
  short* de = demb + (int64_t)b * Fe * D;
  for (int t = lane; t < F * D; t += 64) {
    int i = t / D, d = t % D;
    float acc = 0.0f;
    const short* prow = &plut[i * F];
    const short* fd = f + d;
#pragma unroll 4
    for (int j = 0; j < F; ++j) {
      int p = prow[j];
      if (p < 0) continue;
      acc += bf2f_u16(g[p]) * bf2f_u16(fd[j * D]);
    }
    if (i == 0) {
      // bot row: the direct top-MLP slice adds in
      db[d] = f2bf_u16(acc + bf2f_u16(gdirect[d]));
    } else {
      de[(i - 1) * D + d] = f2bf_u16(acc);
    }
  }
}

__global__ void k_interact_bwd(const short* __restrict__ grad,
                               const short* __restrict__ feats, int B, int F,
                               int D, int P, int P_pad,
                               short* __restrict__ dfeats) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int b = blockIdx.x * 4 + wave;
  extern __shared__ short lds[];
  // per-wave scratch: feats (F*D) + grads (P)
  short* f = lds + wave * (F * D + ((P + 7) & ~7));
  short* g = f + F * D;
  const bool active = b < B;
  if (active) {
    const short* fsrc = feats + (int64_t)b * F * D;
    const short* gsrc = grad + (int64_t)b * P_pad;
    for (int t = lane; t * 8 < F * D; t += 64)
      *reinterpret_cast<bf16x8*>(f + t * 8) =
          *reinterpret_cast<const bf16x8*>(fsrc + t * 8);
    for (int t = lane; t < P; t += 64) g[t] = gsrc[t];
  }
  // per-block pair-index LUT: the triangular index arithmetic in the
  // inner loop cost more ALU than the actual FMAs. ALL threads (active or
  // not) fill it and hit both barriers.
  __shared__ short plut[40 * 40];
  for (int t = threadIdx.x; t < F * F; t += blockDim.x) {
    int i = t / F, j = t % F;
    int lo = i < j ? i : j, hi = i < j ? j : i;
    plut[t] = (short)(i == j ? -1
                             : lo * F - (lo * (lo + 1)) / 2 + (hi - lo - 1));
  }
  __syncthreads();
  if (!active) return;
  short* dst = reinterpret_cast<short*>(dfeats) + (int64_t)b * F * D;
  for (int t = lane; t < F * D; t += 64) {
    int i = t / D, d = t % D;
    float acc = 0.0f;
    const short* prow = &plut[i * F];
    const short* fd = f + d;
#pragma unroll 4
    for (int j = 0; j < F; ++j) {
      int p = prow[j];
      if (p < 0) continue;
      acc += bf2f_u16(g[p]) * bf2f_u16(fd[j * D]);
    }
    dst[t] = f2bf_u16(acc);
  }
}

}  // namespace

// ---------------------- host wrappers ----------------------

static const short* bf_ptr(const torch::Tensor& t) {
  return reinterpret_cast<const short*>(t.data_ptr<at::BFloat16>());
}
static short* bf_ptr_mut(torch::Tensor& t) {
  return reinterpret_cast<short*>(t.data_ptr<at::BFloat16>());
}

torch::Tensor linear_fwd(torch::Tensor x, torch::Tensor w_bf16,
                         torch::Tensor bias, int64_t act,
                         int64_t variant) {
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16 && x.is_contiguous());
  int M = x.size(0), K = x.size(1), N = w_bf16.size(0);
  auto out = torch::empty({M, N}, x.options());
  int tiles_n = (N + 63) / 64;
  const float* bp =
      bias.defined() && bias.numel() ? bias.data_ptr<float>() : nullptr;
  if (variant < 0) variant = 0;  // MT=1 measured fastest across the zoo
  if (variant == 1) {
    int blocks = ((M + 127) / 128) * tiles_n;
    k_linear_fwd_t<2><<<blocks, 256, 0, dense_stream()>>>(
        bf_ptr(x), bf_ptr(w_bf16), bp, M, N, K, (int)act, bf_ptr_mut(out));
  } else {
    int blocks = ((M + 63) / 64) * tiles_n;
    k_linear_fwd_t<1><<<blocks, 256, 0, dense_stream()>>>(
        bf_ptr(x), bf_ptr(w_bf16), bp, M, N, K, (int)act, bf_ptr_mut(out));
  }
  return out;
}

torch::Tensor linear_dx(torch::Tensor g, torch::Tensor w_bf16) {
  int M = g.size(0), N = g.size(1), K = w_bf16.size(1);
  auto dx = torch::empty({M, K}, g.options());
  int tiles_k = (K + 63) / 64;
  if (N >= 128) {
    // wide GEMM-K: strided W-column loads dominate -> LDS-staged variant
    int blocks = ((M + 127) / 128) * tiles_k;
    k_linear_dx_lds<<<blocks, 256, 0, dense_stream()>>>(
        bf_ptr(g), bf_ptr(w_bf16), M, N, K, bf_ptr_mut(dx));
  } else {
    int blocks = ((M + 63) / 64) * tiles_k;
    k_linear_dx<<<blocks, 256, 0, dense_stream()>>>(
        bf_ptr(g), bf_ptr(w_bf16), M, N, K, bf_ptr_mut(dx));
  }
  return dx;
}

static std::tuple<torch::Tensor, torch::Tensor> linear_dw_impl(
    torch::Tensor g, torch::Tensor x, torch::Tensor ws, bool want_bias,
    int64_t variant) {
  int M = g.size(0), N = g.size(1), K = x.size(1);
  // one fill-kernel zero for dW + dbias (cheaper than torch's fill
  // machinery; hipMemsetAsync is avoided: memset nodes recorded during
  // hipGraph capture were observed not to replay)
  int64_t ws_len = (int64_t)N * K + (want_bias ? N : 0);
  k_zero_f32d<<<(int)std::min<int64_t>((ws_len / 4 + 255) / 256, 1024),
                256, 0, dense_stream()>>>(ws.data_ptr<float>(), ws_len);
  auto dw = ws.narrow(0, 0, (int64_t)N * K).view({N, K});
  auto db = want_bias ? ws.narrow(0, (int64_t)N * K, N) : torch::Tensor();
  int m_chunks = (M + 127) / 128;
  float* dbp = want_bias ? db.data_ptr<float>() : nullptr;
  if (variant < 0) variant = N >= 64 ? 2 : 0;
  if (variant == 2) {
    // 64-wide tiles, transposed LDS, multi-chunk register accumulation:
    // pick chunks-per-block so the grid stays ~1024 blocks while the
    // atomic flush volume drops by cpb x
    int tiles = ((N + 63) / 64) * ((K + 63) / 64);
    int cpb = std::max(1, (int)((int64_t)m_chunks * tiles / 512));
    int chunk_blocks = (m_chunks + cpb - 1) / cpb;
    k_linear_dw_lds64t<<<tiles * chunk_blocks, 256, 0, dense_stream()>>>(
        bf_ptr(g), bf_ptr(x), M, N, K, cpb, dw.data_ptr<float>(), dbp);
  } else if (variant == 1) {
    int tiles = ((N + 63) / 64) * ((K + 63) / 64);
    k_linear_dw_lds64<<<tiles * m_chunks, 256, 0, dense_stream()>>>(
        bf_ptr(g), bf_ptr(x), M, N, K, dw.data_ptr<float>(), dbp);
  } else {
    int tiles = ((N + 15) / 16) * ((K + 63) / 64);
    k_linear_dw_lds<<<tiles * m_chunks, 256, 0, dense_stream()>>>(
        bf_ptr(g), bf_ptr(x), M, N, K, dw.data_ptr<float>(), dbp);
  }
  return {dw, db};
}

std::tuple<torch::Tensor, torch::Tensor> linear_dw(torch::Tensor g,
                                                   torch::Tensor x,
                                                   bool want_bias,
                                                   int64_t variant) {
  int N = g.size(1), K = x.size(1);
  int64_t ws_len = (int64_t)N * K + (want_bias ? N : 0);
  auto ws = torch::empty({ws_len}, g.options().dtype(torch::kFloat32));
  return linear_dw_impl(g, x, ws, want_bias, variant);
}

// dW/db written into a caller-provided flat [N*K + N] workspace — a view
// of the optimizer's flat gradient buffer, so backward lands gradients
// exactly where the fused dense Adam reads them (no flatten/unflatten
// copies around the all-reduce).
std::tuple<torch::Tensor, torch::Tensor> linear_dw_out(torch::Tensor g,
                                                       torch::Tensor x,
                                                       torch::Tensor ws,
                                                       bool want_bias,
                                                       int64_t variant) {
  TORCH_CHECK(ws.is_contiguous() && ws.scalar_type() == torch::kFloat32);
  int N = g.size(1), K = x.size(1);
  int64_t ws_len = (int64_t)N * K + (want_bias ? N : 0);
  TORCH_CHECK(ws.numel() == ws_len, "linear_dw_out: workspace size");
  return linear_dw_impl(g, x, ws, want_bias, variant);
}

std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> resln_fwd(
    torch::Tensor x, torch::Tensor a, torch::Tensor gamma,
    torch::Tensor beta, double eps) {
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16 && x.is_contiguous());
  int64_t M = x.numel() / x.size(-1);
  int N = x.size(-1);
  TORCH_CHECK(N <= 64, "resln: row width must be <= 64");
  auto y = torch::empty_like(x);
  auto z = torch::empty_like(x);
  auto stats = torch::empty({M, 2}, x.options().dtype(torch::kFloat32));
  if (M == 0) return {y, z, stats};
  TORCH_CHECK(N == 16 || N == 32 || N == 64, "resln: N must be 16/32/64");
  int blocks = (int)std::min<int64_t>((M + 255) / 256, 4096);
#define RESLN_FWD(NV) \
  k_resln_fwd<NV><<<blocks, 256, 0, dense_stream()>>>( \
      bf_ptr(x), bf_ptr(a), gamma.data_ptr<float>(), \
      beta.data_ptr<float>(), M, (float)eps, bf_ptr_mut(y), \
      bf_ptr_mut(z), stats.data_ptr<float>())
  if (N == 16) RESLN_FWD(16);
  else if (N == 32) RESLN_FWD(32);
  else RESLN_FWD(64);
#undef RESLN_FWD
  return {y, z, stats};
}

std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> resln_bwd(
    torch::Tensor dy, torch::Tensor z, torch::Tensor stats,
    torch::Tensor gamma) {
  int64_t M = z.numel() / z.size(-1);
  int N = z.size(-1);
  auto dz = torch::empty_like(z);
  auto opts = gamma.options();
  auto dgamma = torch::empty({N}, opts);
  auto dbeta = torch::empty({N}, opts);
  auto stream = dense_stream();
  k_zero_f32d<<<1, 64, 0, stream>>>(dgamma.data_ptr<float>(), N);
  k_zero_f32d<<<1, 64, 0, stream>>>(dbeta.data_ptr<float>(), N);
  if (M == 0) return {dz, dgamma, dbeta};
  auto dyc = dy.contiguous();
  // ~8 rows per thread amortize the dgamma/dbeta partial merges
  int blocks = (int)std::min<int64_t>((M + 256 * 8 - 1) / (256 * 8), 4096);
  blocks = std::max(blocks, 64);
#define RESLN_BWD(NV) \
  k_resln_bwd<NV><<<blocks, 256, 0, stream>>>( \
      bf_ptr(dyc), bf_ptr(z), stats.data_ptr<float>(), \
      gamma.data_ptr<float>(), M, bf_ptr_mut(dz), \
      dgamma.data_ptr<float>(), dbeta.data_ptr<float>())
  if (N == 16) RESLN_BWD(16);
  else if (N == 32) RESLN_BWD(32);
  else RESLN_BWD(64);
#undef RESLN_BWD
  return {dz, dgamma, dbeta};
}

void dense_adam(torch::Tensor w, torch::Tensor g, torch::Tensor m,
                torch::Tensor v, torch::Tensor w16, torch::Tensor powers,
                double lr, double beta1, double beta2, double eps,
                double gscale) {
  int64_t n = w.numel();
  if (n == 0) return;
  short* w16p = (w16.defined() && w16.numel())
                    ? reinterpret_cast<short*>(w16.data_ptr<at::BFloat16>())
                    : nullptr;
  int blocks = (int)std::min<int64_t>((n + 255) / 256, 4096);
  k_dense_adam<<<blocks, 256, 0, dense_stream()>>>(
      w.data_ptr<float>(), g.data_ptr<float>(), m.data_ptr<float>(),
      v.data_ptr<float>(), w16p, powers.data_ptr<float>(), n, (float)lr,
      (float)beta1, (float)beta2, (float)eps, (float)gscale);
}

torch::Tensor act_bwd(torch::Tensor dy, torch::Tensor out, int64_t act) {
  auto g = torch::empty_like(dy);
  int64_t n = dy.numel();
  int blocks = (int)std::min<int64_t>((n / 8 + 255) / 256, 4096);
  k_act_bwd<<<std::max(blocks, 1), 256, 0, dense_stream()>>>(
      bf_ptr(dy), bf_ptr(out), n, (int)act, bf_ptr_mut(g));
  return g;
}

torch::Tensor interact_fwd(torch::Tensor feats, int64_t p_pad) {
  TORCH_CHECK(feats.scalar_type() == torch::kBFloat16 &&
              feats.is_contiguous());
  int B = feats.size(0), F = feats.size(1), D = feats.size(2);
  int P = F * (F - 1) / 2;
  TORCH_CHECK((F * D) % 8 == 0, "F*D must be a multiple of 8");
  auto out = torch::empty({B, p_pad}, feats.options());
  int blocks = (B + 3) / 4;
  size_t lds = 4 * (size_t)F * D * sizeof(short);
  k_interact_fwd<<<blocks, 256, lds, dense_stream()>>>(
      bf_ptr(feats), B, F, D, P, (int)p_pad, bf_ptr_mut(out));
  return out;
}

torch::Tensor interact_bwd(torch::Tensor grad, torch::Tensor feats) {
  int B = feats.size(0), F = feats.size(1), D = feats.size(2);
  TORCH_CHECK(F <= 40, "interact_bwd: pair LUT supports F <= 40");
  int P = F * (F - 1) / 2;
  int P_pad = grad.size(1);
  auto dfeats = torch::empty_like(feats);
  int blocks = (B + 3) / 4;
  size_t lds = 4 * (size_t)(F * D + ((P + 7) & ~7)) * sizeof(short);
  k_interact_bwd<<<blocks, 256, lds, dense_stream()>>>(
      bf_ptr(grad.contiguous()), bf_ptr(feats), B, F, D, P, P_pad,
      bf_ptr_mut(dfeats));
  return dfeats;
}

torch::Tensor interact_cat_fwd(torch::Tensor bot, torch::Tensor emb,
                               int64_t p_pad) {
  TORCH_CHECK(bot.scalar_type() == torch::kBFloat16 && bot.is_contiguous());
  TORCH_CHECK(emb.scalar_type() == torch::kBFloat16 && emb.is_contiguous());
  int B = emb.size(0), Fe = emb.size(1), D = emb.size(2);
  int F = Fe + 1;
  int P = F * (F - 1) / 2;
  TORCH_CHECK(D % 8 == 0, "D must be a multiple of 8");
  auto out = torch::empty({B, D + p_pad}, emb.options());
  int blocks = (B + 3) / 4;
  size_t lds = 4 * (size_t)F * D * sizeof(short);
  k_interact_cat_fwd<<<blocks, 256, lds, dense_stream()>>>(
      bf_ptr(bot), bf_ptr(emb), B, Fe, D, P, (int)p_pad, bf_ptr_mut(out));
  return out;
}

std::tuple<torch::Tensor, torch::Tensor> interact_cat_bwd(
    torch::Tensor grad, torch::Tensor bot, torch::Tensor emb) {
  int B = emb.size(0), Fe = emb.size(1), D = emb.size(2);
  int F = Fe + 1;
  TORCH_CHECK(F <= 40, "interact: pair LUT supports F <= 40");
  int P = F * (F - 1) / 2;
  int P_pad = grad.size(1) - D;
  auto dbot = torch::empty_like(bot);
  auto demb = torch::empty_like(emb);
  int blocks = (B + 3) / 4;
  size_t lds = 4 * (size_t)(F * D + ((P + 7) & ~7)) * sizeof(short);
  k_interact_cat_bwd<<<blocks, 256, lds, dense_stream()>>>(
      bf_ptr(grad.contiguous()), bf_ptr(bot), bf_ptr(emb), B, Fe, D, P,
      P_pad, bf_ptr_mut(dbot), bf_ptr_mut(demb));
  return {dbot, demb};
}

std::tuple<torch::Tensor, torch::Tensor> l2norm_fwd(torch::Tensor x,
                                                    double eps) {
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16 && x.is_contiguous());
  int M = x.size(0), N = x.size(1);
  auto y = torch::empty_like(x);
  auto inv = torch::empty({M}, x.options().dtype(torch::kFloat32));
  k_l2norm_fwd<<<(M + 3) / 4, 256, 0, dense_stream()>>>(
      bf_ptr(x), M, N, (float)eps, bf_ptr_mut(y), inv.data_ptr<float>());
  return {y, inv};
}

torch::Tensor l2norm_bwd(torch::Tensor dy, torch::Tensor y,
                         torch::Tensor inv) {
  int M = y.size(0), N = y.size(1);
  auto dx = torch::empty_like(y);
  k_l2norm_bwd<<<(M + 3) / 4, 256, 0, dense_stream()>>>(
      bf_ptr(dy.contiguous()), bf_ptr(y), inv.data_ptr<float>(), M, N,
      bf_ptr_mut(dx));
  return dx;
}

void register_dense(py::module_& mod) {
  mod.def("l2norm_fwd", &l2norm_fwd);
  mod.def("l2norm_bwd", &l2norm_bwd);
  mod.def("interact_fwd", &interact_fwd);
  mod.def("interact_bwd", &interact_bwd);
  mod.def("interact_cat_fwd", &interact_cat_fwd);
  mod.def("interact_cat_bwd", &interact_cat_bwd);
  mod.def("linear_fwd", &linear_fwd, py::arg("x"), py::arg("w"),
          py::arg("bias"), py::arg("act"), py::arg("variant") = -1);
  mod.def("linear_dx", &linear_dx);
  mod.def("linear_dw", &linear_dw, py::arg("g"), py::arg("x"),
          py::arg("want_bias"), py::arg("variant") = -1);
  mod.def("linear_dw_out", &linear_dw_out, py::arg("g"), py::arg("x"),
          py::arg("ws"), py::arg("want_bias"), py::arg("variant") = -1);
  mod.def("dense_adam", &dense_adam);
  mod.def("resln_fwd", &resln_fwd);
  mod.def("resln_bwd", &resln_bwd);
  mod.def("act_bwd", &act_bwd);
}
