// fp8_kernels.hip — OCP fp8 (e4m3fn) inference path for gfx950.
//
// The reference ships a post-training low-precision tool
// (tools/low_precision_optimize/, int8/fp16 over embeddings + dense);
// the MI355X-native angle is OCP fp8: CDNA4 multiplies non-scaled e4m3
// at the full bf16 MFMA rate (v_mfma_f32_16x16x32_fp8_fp8), so an fp8
// serving MLP halves weight/activation bytes and L2 footprint at zero
// math-rate cost. The 2x math-rate form (MX block-scaled
// mfma_scale_f32_16x16x128_f8f6f4, K=128 with per-32-element scales) is
// deliberately NOT used here: at DLRM MLP shapes the forward is
// latency-bound (see profiles/PERF_LOG.md round-2 ladder), so the win
// to harvest is bytes, not flops.
//
// Scheme: per-row dynamic scales (amax/448 — 448 = e4m3fn max finite).
//   activations: quantized on the fly per batch row;
//   weights:     quantized once at conversion per output channel
//                (a row of torch's [N, K] weight layout).
//   y[m,n] = sa[m]*sw[n] * (qA[m,:] . qW[n,:]) + bias[n]   (bf16 out)
// Accumulation is fp32 inside the MFMA, products of e4m3 values are
// exact in fp32 — the only quantization error is the e4m3 rounding of
// the inputs, which tests bound against a torch fp32 reference.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp8.h>

#include <cstdint>

namespace py = pybind11;

static inline hipStream_t fp8_stream() {
  return at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
}

namespace {

using f32x4 = __attribute__((ext_vector_type(4))) float;

__device__ __forceinline__ float fp8_bf2f(short u) {
  union { unsigned int i; float f; } cv;
  cv.i = ((unsigned int)(unsigned short)u) << 16;
  return cv.f;
}

__device__ __forceinline__ short fp8_f2bf(float f) {
  union { float f; unsigned int i; } cv;
  cv.f = f;
  unsigned int lsb = (cv.i >> 16) & 1;
  cv.i += 0x7fff + lsb;
  return (short)(cv.i >> 16);
}

__device__ __forceinline__ uint8_t f32_to_e4m3(float v) {
  // explicit clamp so the kernel and the torch reference agree on the
  // boundary (SATFINITE also clamps, but clamping BEFORE the convert
  // keeps both sides in the exactly-representable range)
  v = fminf(fmaxf(v, -448.0f), 448.0f);
  return (uint8_t)__hip_cvt_float_to_fp8(v, __HIP_SATFINITE, __HIP_E4M3);
}

// ------------------------------------------------------------------
// Per-row e4m3 quantization: q[r, :] = rne(x[r, :] / s_r) with
// s_r = amax_r / 448 (s_r = 1 for all-zero rows). One block per row,
// grid-stride over rows; input fp32 or bf16 (in_f XOR in_b non-null).
// ------------------------------------------------------------------
__global__ void k_quant_rows_e4m3(const float* __restrict__ in_f,
                                  const short* __restrict__ in_b,
                                  int64_t n, int d,
                                  uint8_t* __restrict__ q,
                                  float* __restrict__ scale) {
  __shared__ float red[256];
  for (int64_t row = blockIdx.x; row < n; row += gridDim.x) {
    const int64_t base = row * d;
    float am = 0.0f;
    for (int i = threadIdx.x; i < d; i += blockDim.x) {
      float v = in_f ? in_f[base + i] : fp8_bf2f(in_b[base + i]);
      am = fmaxf(am, fabsf(v));
    }
    red[threadIdx.x] = am;
    __syncthreads();
    for (int off = blockDim.x >> 1; off > 0; off >>= 1) {
      if (threadIdx.x < off)
        red[threadIdx.x] = fmaxf(red[threadIdx.x], red[threadIdx.x + off]);
      __syncthreads();
    }
    const float amax = red[0];
    const float s = amax > 0.0f ? amax / 448.0f : 1.0f;
    // true divide, not reciprocal-multiply: keeps the scaled value
    // bit-identical to the torch reference (a 1-ulp fp32 difference at
    // an e4m3 rounding tie moves the byte by a whole e4m3 ulp)
    for (int i = threadIdx.x; i < d; i += blockDim.x) {
      float v = in_f ? in_f[base + i] : fp8_bf2f(in_b[base + i]);
      q[base + i] = f32_to_e4m3(v / s);
    }
    if (threadIdx.x == 0) scale[row] = s;
    __syncthreads();  // red[] reused by the next grid-stride row
  }
}

// 8 consecutive e4m3 bytes of a row-major [rows, cols] matrix as the
// i64 MFMA fragment payload; ZERO-filled outside bounds (e4m3 0x00 = 0).
__device__ __forceinline__ long load_frag8_e4m3(
    const uint8_t* __restrict__ src, int r, int k0, int rows, int cols) {
  if (r < rows && k0 + 7 < cols) {
    long v;
    __builtin_memcpy(&v, src + (int64_t)r * cols + k0, 8);
    return v;
  }
  union { long l; uint8_t b[8]; } u;
  u.l = 0;
  if (r < rows) {
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      int k = k0 + i;
      if (k < cols) u.b[i] = src[(int64_t)r * cols + k];
    }
  }
  return u.l;
}

// ------------------------------------------------------------------
// C[M,N] = act(sa[m]*sw[n] * (qA[M,K] @ qW[N,K]^T) + bias); bf16 out.
// Same tiling as the bf16 k_linear_fwd_t (dense_kernels.hip): 4 waves
// stacked on M, 16x64 tile per wave, mfma_f32_16x16x32_fp8_fp8 with the
// 8-elements-per-lane fragment mapping (A: row=lane%16, k=8*(lane/16)+i;
// B: col=lane%16, same k) — verified on-device against a torch fp32
// reference with asymmetric random inputs (tests/test_gpu_fp8.py).
// ------------------------------------------------------------------
__global__ void k_linear_fwd_fp8(const uint8_t* __restrict__ A,
                                 const float* __restrict__ sa,
                                 const uint8_t* __restrict__ W,
                                 const float* __restrict__ sw,
                                 const float* __restrict__ bias,
                                 int M, int N, int K, int act,
                                 short* __restrict__ C) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int ntiles64 = (N + 63) / 64;
  const int m0 = (blockIdx.x / ntiles64) * 64 + wave * 16;
  const int n0 = (blockIdx.x % ntiles64) * 64;
  if (m0 >= M) return;
  const int row_a = m0 + (lane & 15);
  const int col_b = n0 + (lane & 15);
  const int kgrp = (lane >> 4) * 8;
  f32x4 acc[4] = {};
  const int nt = min(4, (N - n0 + 15) / 16);
  for (int k0 = 0; k0 < K; k0 += 32) {
    long a = load_frag8_e4m3(A, row_a, k0 + kgrp, M, K);
#pragma unroll
    for (int t = 0; t < 4; ++t) {
      if (t >= nt) break;
      long b = load_frag8_e4m3(W, col_b + t * 16, k0 + kgrp, N, K);
      acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(a, b, acc[t],
                                                          0, 0, 0);
    }
  }
#pragma unroll
  for (int t = 0; t < 4; ++t) {
    if (t >= nt) break;
    const int cn = n0 + t * 16 + (lane & 15);
    if (cn >= N) continue;
    const float swv = sw[cn];
    const float bv = bias ? bias[cn] : 0.0f;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int cm = m0 + (lane >> 4) * 4 + i;
      if (cm >= M) continue;
      float v = acc[t][i] * sa[cm] * swv + bv;
      if (act == 1 && v < 0.0f) v = 0.0f;
      else if (act == 2) v = 1.0f / (1.0f + __expf(-v));
      C[(int64_t)cm * N + cn] = fp8_f2bf(v);
    }
  }
}

}  // namespace

// ------------------------------------------------------------------
// wrappers
// ------------------------------------------------------------------

std::tuple<torch::Tensor, torch::Tensor> quant_rows_e4m3(torch::Tensor x) {
  TORCH_CHECK(x.dim() == 2 && x.is_contiguous(), "x must be 2-D contiguous");
  TORCH_CHECK(x.scalar_type() == torch::kFloat32
                  || x.scalar_type() == torch::kBFloat16,
              "x must be fp32 or bf16");
  const int64_t n = x.size(0);
  const int d = (int)x.size(1);
  auto q = torch::empty({n, (int64_t)d}, x.options().dtype(torch::kUInt8));
  auto s = torch::empty({n}, x.options().dtype(torch::kFloat32));
  if (n == 0) return {q, s};
  const int blocks = (int)std::min<int64_t>(n, 16384);
  const float* pf = nullptr;
  const short* pb = nullptr;
  if (x.scalar_type() == torch::kFloat32) pf = x.data_ptr<float>();
  else pb = reinterpret_cast<const short*>(x.data_ptr<at::BFloat16>());
  k_quant_rows_e4m3<<<blocks, 256, 0, fp8_stream()>>>(
      pf, pb, n, d, q.data_ptr<uint8_t>(), s.data_ptr<float>());
  return {q, s};
}

torch::Tensor linear_fwd_fp8(torch::Tensor qx, torch::Tensor sx,
                             torch::Tensor qw, torch::Tensor sw,
                             c10::optional<torch::Tensor> bias,
                             int64_t act) {
  TORCH_CHECK(qx.scalar_type() == torch::kUInt8 && qx.is_contiguous());
  TORCH_CHECK(qw.scalar_type() == torch::kUInt8 && qw.is_contiguous());
  TORCH_CHECK(qx.size(1) == qw.size(1), "K mismatch");
  const int M = (int)qx.size(0), K = (int)qx.size(1), N = (int)qw.size(0);
  TORCH_CHECK(sx.numel() == M && sw.numel() == N, "scale shape mismatch");
  auto C = torch::empty({(int64_t)M, (int64_t)N},
                        qx.options().dtype(torch::kBFloat16));
  if (M == 0) return C;
  const float* bp = nullptr;
  if (bias.has_value() && bias->defined()) {
    TORCH_CHECK(bias->scalar_type() == torch::kFloat32
                    && bias->numel() == N, "bias must be fp32 [N]");
    bp = bias->data_ptr<float>();
  }
  const int mtiles = (M + 63) / 64, ntiles = (N + 63) / 64;
  k_linear_fwd_fp8<<<mtiles * ntiles, 256, 0, fp8_stream()>>>(
      qx.data_ptr<uint8_t>(), sx.data_ptr<float>(),
      qw.data_ptr<uint8_t>(), sw.data_ptr<float>(), bp, M, N, K, (int)act,
      reinterpret_cast<short*>(C.data_ptr<at::BFloat16>()));
  return C;
}

void register_fp8(py::module_& mod) {
  mod.def("quant_rows_e4m3", &quant_rows_e4m3);
  mod.def("linear_fwd_fp8", &linear_fwd_fp8);
}
