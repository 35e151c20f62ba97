// attention_kernels.hip — fused multi-head attention for short behavior
// sequences (MI355X / gfx950).
//
// The sequence models (DIN/BST, reference modelzoo/bst + din attention
// blocks) attend over T ≈ 50-100 items at d_model ≈ 32-64 — shapes where
// library attention (batched GEMMs + separate softmax/mask kernels) is
// pure launch overhead. CDNA4-first design instead: ONE workgroup per
// sample holds the whole sequence in LDS (K/V ≈ a few KB at these
// shapes), each thread owns one (query row, head) pair and produces its
// output row without any inter-thread communication, and the backward
// recomputes softmax from saved per-row (max, sum) stats instead of
// materializing [B, h, T, T] probabilities (170 MB at batch 8192).
//
// Capability ≙ the reference's transformer/attention blocks
// (modelzoo/bst/train.py, modelzoo/din/train.py:207-253 attention); the
// kernel design is new for this engine.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <cstdint>

namespace {

static inline hipStream_t att_stream() {
  return at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
}

__device__ __forceinline__ float bfu(short u) {
  unsigned int x = ((unsigned int)(unsigned short)u) << 16;
  return __uint_as_float(x);
}
__device__ __forceinline__ short fbu(float f) {
  // round-to-nearest-even bf16
  unsigned int x = __float_as_uint(f);
  unsigned int r = x + 0x7FFF + ((x >> 16) & 1);
  return (short)(r >> 16);
}

constexpr int kMaxDh = 32;  // head dim cap (template dispatch below)

// Forward: one workgroup per sample. Threads cover (q_row, head) pairs.
// q/k/v: [B, T, D] bf16; key_pad: [B, T] uint8 (1 = masked key);
// out: [B, T, D] bf16; stats: [B, h, T, 2] fp32 (row max, row exp-sum).
// DH is a compile-time head dim so the per-thread accumulators are
// exact-size register arrays with fully unrolled loops (a kMaxDh-sized
// runtime-bounded array cost ~200 VGPRs and 4x the runtime).
template <int DH>
__global__ void k_mha_fwd(const short* __restrict__ q,
                          const short* __restrict__ k,
                          const short* __restrict__ v,
                          const uint8_t* __restrict__ key_pad, int B, int T,
                          int D, int H, float scale,
                          short* __restrict__ out,
                          float* __restrict__ stats) {
  extern __shared__ short lds[];
  short* K = lds;            // [T, D]
  short* V = lds + T * D;    // [T, D]
  const int b = blockIdx.x;
  constexpr int dh = DH;
  const int64_t base = (int64_t)b * T * D;
  for (int i = threadIdx.x; i < T * D; i += blockDim.x) {
    K[i] = k[base + i];
    V[i] = v[base + i];
  }
  __syncthreads();
  const uint8_t* pad = key_pad + (int64_t)b * T;
  for (int pair = threadIdx.x; pair < T * H; pair += blockDim.x) {
    const int qr = pair / H;
    const int h = pair % H;
    float qv[DH];
    const short* qp = q + base + (int64_t)qr * D + h * dh;
#pragma unroll
    for (int d = 0; d < dh; ++d) qv[d] = bfu(qp[d]);
    // pass 1: row max
    float mx = -1e30f;
    for (int t = 0; t < T; ++t) {
      if (pad[t]) continue;
      const short* kp = K + t * D + h * dh;
      float s = 0.0f;
      for (int d = 0; d < dh; ++d) s += qv[d] * bfu(kp[d]);
      s *= scale;
      mx = fmaxf(mx, s);
    }
    // pass 2: exp-sum + weighted V accumulation
    float acc[DH];
#pragma unroll
    for (int d = 0; d < dh; ++d) acc[d] = 0.0f;
    float sum = 0.0f;
    for (int t = 0; t < T; ++t) {
      if (pad[t]) continue;
      const short* kp = K + t * D + h * dh;
      float s = 0.0f;
      for (int d = 0; d < dh; ++d) s += qv[d] * bfu(kp[d]);
      float p = __expf(s * scale - mx);
      sum += p;
      const short* vp = V + t * D + h * dh;
      for (int d = 0; d < dh; ++d) acc[d] += p * bfu(vp[d]);
    }
    const float inv = 1.0f / fmaxf(sum, 1e-20f);
    short* op = out + base + (int64_t)qr * D + h * dh;
    for (int d = 0; d < dh; ++d) op[d] = fbu(acc[d] * inv);
    float* st = stats + (((int64_t)b * H + h) * T + qr) * 2;
    st[0] = mx;
    st[1] = sum;
  }
}

// Backward: recompute probabilities from stats; two phases per block.
// Phase A (q rows): dQ + per-row dot sum_t p*dP. Phase B (k rows):
// dK, dV. Each thread owns one output row — no atomics anywhere.
template <int DH>
__global__ void k_mha_bwd(const short* __restrict__ dout,
                          const short* __restrict__ q,
                          const short* __restrict__ k,
                          const short* __restrict__ v,
                          const uint8_t* __restrict__ key_pad,
                          const float* __restrict__ stats, int B, int T,
                          int D, int H, float scale,
                          short* __restrict__ dq, short* __restrict__ dk,
                          short* __restrict__ dv) {
  extern __shared__ short lds[];
  short* Q = lds;                       // [T, D]
  short* K = lds + T * D;               // [T, D]
  short* V = lds + 2 * T * D;           // [T, D]
  short* dO = lds + 3 * T * D;          // [T, D]
  float* row_dot = reinterpret_cast<float*>(lds + 4 * T * D);  // [T*H]
  const int b = blockIdx.x;
  constexpr int dh = DH;
  const int64_t base = (int64_t)b * T * D;
  for (int i = threadIdx.x; i < T * D; i += blockDim.x) {
    Q[i] = q[base + i];
    K[i] = k[base + i];
    V[i] = v[base + i];
    dO[i] = dout[base + i];
  }
  __syncthreads();
  const uint8_t* pad = key_pad + (int64_t)b * T;
  const float* st_b = stats + (int64_t)b * H * T * 2;
  // phase A: per (q, head): row_dot and dQ
  for (int pair = threadIdx.x; pair < T * H; pair += blockDim.x) {
    const int qr = pair / H;
    const int h = pair % H;
    const float* st = st_b + ((int64_t)h * T + qr) * 2;
    const float mx = st[0];
    const float inv = 1.0f / fmaxf(st[1], 1e-20f);
    float qv[DH], go[DH];
#pragma unroll
    for (int d = 0; d < dh; ++d) {
      qv[d] = bfu(Q[qr * D + h * dh + d]);
      go[d] = bfu(dO[qr * D + h * dh + d]);
    }
    float rd = 0.0f;
    for (int t = 0; t < T; ++t) {
      if (pad[t]) continue;
      float s = 0.0f, dp = 0.0f;
      const short* kp = K + t * D + h * dh;
      const short* vp = V + t * D + h * dh;
      for (int d = 0; d < dh; ++d) {
        s += qv[d] * bfu(kp[d]);
        dp += go[d] * bfu(vp[d]);
      }
      rd += __expf(s * scale - mx) * inv * dp;
    }
    row_dot[qr * H + h] = rd;
    float dqv[DH];
#pragma unroll
    for (int d = 0; d < dh; ++d) dqv[d] = 0.0f;
    for (int t = 0; t < T; ++t) {
      if (pad[t]) continue;
      float s = 0.0f, dp = 0.0f;
      const short* kp = K + t * D + h * dh;
      const short* vp = V + t * D + h * dh;
      for (int d = 0; d < dh; ++d) {
        s += qv[d] * bfu(kp[d]);
        dp += go[d] * bfu(vp[d]);
      }
      float p = __expf(s * scale - mx) * inv;
      float ds = p * (dp - rd) * scale;
      for (int d = 0; d < dh; ++d) dqv[d] += ds * bfu(kp[d]);
    }
    short* dqp = dq + base + (int64_t)qr * D + h * dh;
    for (int d = 0; d < dh; ++d) dqp[d] = fbu(dqv[d]);
  }
  __syncthreads();
  // phase B: per (key row, head): dK, dV
  for (int pair = threadIdx.x; pair < T * H; pair += blockDim.x) {
    const int t = pair / H;
    const int h = pair % H;
    short* dkp = dk + base + (int64_t)t * D + h * dh;
    short* dvp = dv + base + (int64_t)t * D + h * dh;
    if (pad[t]) {
      for (int d = 0; d < dh; ++d) {
        dkp[d] = 0;
        dvp[d] = 0;
      }
      continue;
    }
    float kv[DH], vv[DH], dkv[DH], dvv[DH];
#pragma unroll
    for (int d = 0; d < dh; ++d) {
      kv[d] = bfu(K[t * D + h * dh + d]);
      vv[d] = bfu(V[t * D + h * dh + d]);
      dkv[d] = 0.0f;
      dvv[d] = 0.0f;
    }
    for (int qr = 0; qr < T; ++qr) {
      const float* st = st_b + ((int64_t)h * T + qr) * 2;
      const float mx = st[0];
      const float inv = 1.0f / fmaxf(st[1], 1e-20f);
      float s = 0.0f, dp = 0.0f;
      const short* qp = Q + qr * D + h * dh;
      const short* gp = dO + qr * D + h * dh;
      for (int d = 0; d < dh; ++d) {
        s += bfu(qp[d]) * kv[d];
        dp += bfu(gp[d]) * vv[d];
      }
      float p = __expf(s * scale - mx) * inv;
      float ds = p * (dp - row_dot[qr * H + h]) * scale;
      for (int d = 0; d < dh; ++d) {
        dkv[d] += ds * bfu(qp[d]);
        dvv[d] += p * bfu(gp[d]);
      }
    }
    for (int d = 0; d < dh; ++d) {
      dkp[d] = fbu(dkv[d]);
      dvp[d] = fbu(dvv[d]);
    }
  }
}

// ---------------------------------------------------------------------
// DIN attention features: att_in[b,t] = [s, tgt, s - tgt, s * tgt]
// built in ONE pass (the torch path materializes the expand, two
// elementwise intermediates and a 4-way cat — 5 kernels and ~2x the
// HBM traffic at [B, T, 4D]). ≙ modelzoo/din/train.py attention input.
__global__ void k_din_feat_fwd(const float* __restrict__ seq,
                               const float* __restrict__ tgt, int B, int T,
                               int D, short* __restrict__ out) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t total = (int64_t)B * T * D;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; i < total; i += stride) {
    int d = (int)(i % D);
    int64_t bt = i / D;
    int b = (int)(bt / T);
    float s = seq[i];
    float t = tgt[(int64_t)b * D + d];
    short* o = out + bt * 4 * D + d;
    o[0] = fbu(s);
    o[D] = fbu(t);
    o[2 * D] = fbu(s - t);
    o[3 * D] = fbu(s * t);
  }
}

// backward: ds = g0 + g2 + g3*t (elementwise); dt[b] = sum_t
// (g1 - g2 + g3*s) via one thread per (b, d) looping T — NO atomics
// (the per-element atomicAdd variant measured 132 us/step at B*T*D
// 13M atomics).
__global__ void k_din_feat_bwd_ds(const short* __restrict__ g,
                                  const float* __restrict__ tgt, int B,
                                  int T, int D,
                                  float* __restrict__ dseq) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t total = (int64_t)B * T * D;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; i < total; i += stride) {
    int d = (int)(i % D);
    int64_t bt = i / D;
    int b = (int)(bt / T);
    const short* gp = g + bt * 4 * D + d;
    dseq[i] = bfu(gp[0]) + bfu(gp[2 * D])
              + bfu(gp[3 * D]) * tgt[(int64_t)b * D + d];
  }
}

__global__ void k_din_feat_bwd_dt(const short* __restrict__ g,
                                  const float* __restrict__ seq, int B,
                                  int T, int D,
                                  float* __restrict__ dtgt) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t total = (int64_t)B * D;
  int64_t stride = gridDim.x * (int64_t)blockDim.x;
  for (; i < total; i += stride) {
    int d = (int)(i % D);
    int b = (int)(i / D);
    float acc = 0.0f;
    for (int t = 0; t < T; ++t) {
      int64_t bt = (int64_t)b * T + t;
      const short* gp = g + bt * 4 * D + d;
      acc += bfu(gp[D]) - bfu(gp[2 * D])
             + bfu(gp[3 * D]) * seq[bt * D + d];
    }
    dtgt[i] = acc;
  }
}

// ------------------------------------------------------------------
// DIN attention tail: masked softmax over scores [B, T] + weighted
// pooling of seq [B, T, D] fused into ONE kernel per direction. The
// torch chain (masked_fill, softmax, [B,T,D] broadcast-mul
// materialization, sum, + their backwards) was ~8 launches and the
// dominant torch-glue cost of the DIN step (prof_din2.csv). A fully
// masked row pools to zeros (matches "masked positions train nothing").
// ------------------------------------------------------------------
__global__ void k_msm_pool_fwd(const float* __restrict__ scores,
                               const float* __restrict__ seq,
                               const uint8_t* __restrict__ mask,
                               int B, int T, int D,
                               float* __restrict__ w_out,
                               float* __restrict__ out) {
  extern __shared__ float smw[];  // [T]
  __shared__ float red[256];
  const int b = blockIdx.x;
  if (b >= B) return;
  float lmax = -3.4e38f;
  for (int t = threadIdx.x; t < T; t += blockDim.x) {
    float s = mask[b * T + t] ? scores[b * T + t] : -3.4e38f;
    smw[t] = s;
    lmax = fmaxf(lmax, s);
  }
  red[threadIdx.x] = lmax;
  __syncthreads();
  for (int off = blockDim.x >> 1; off > 0; off >>= 1) {
    if (threadIdx.x < off)
      red[threadIdx.x] = fmaxf(red[threadIdx.x], red[threadIdx.x + off]);
    __syncthreads();
  }
  const float m = red[0];
  __syncthreads();
  float lsum = 0.0f;
  for (int t = threadIdx.x; t < T; t += blockDim.x) {
    float e = (smw[t] <= -3.0e38f) ? 0.0f : __expf(smw[t] - m);
    smw[t] = e;
    lsum += e;
  }
  red[threadIdx.x] = lsum;
  __syncthreads();
  for (int off = blockDim.x >> 1; off > 0; off >>= 1) {
    if (threadIdx.x < off) red[threadIdx.x] += red[threadIdx.x + off];
    __syncthreads();
  }
  const float inv = 1.0f / (red[0] + 1e-20f);
  __syncthreads();
  for (int t = threadIdx.x; t < T; t += blockDim.x) {
    smw[t] *= inv;
    w_out[b * T + t] = smw[t];
  }
  __syncthreads();
  for (int d = threadIdx.x; d < D; d += blockDim.x) {
    float acc = 0.0f;
    const float* sp = seq + (int64_t)b * T * D + d;
    for (int t = 0; t < T; ++t) acc += smw[t] * sp[(int64_t)t * D];
    out[(int64_t)b * D + d] = acc;
  }
}

// dscores[t] = mask ? w[t] * (dout.seq[t] - sum_u w[u] (dout.seq[u])) : 0
// dseq[t, d] = w[t] * dout[d]
__global__ void k_msm_pool_bwd(const float* __restrict__ dout,
                               const float* __restrict__ seq,
                               const float* __restrict__ w,
                               const uint8_t* __restrict__ mask,
                               int B, int T, int D,
                               float* __restrict__ dscores,
                               float* __restrict__ dseq) {
  extern __shared__ float sdot[];  // [T] row dots, then [D] dout row
  __shared__ float red[256];
  const int b = blockIdx.x;
  if (b >= B) return;
  float* sdo = sdot + T;
  for (int d = threadIdx.x; d < D; d += blockDim.x)
    sdo[d] = dout[(int64_t)b * D + d];
  __syncthreads();
  for (int t = threadIdx.x; t < T; t += blockDim.x) {
    const float* sp = seq + ((int64_t)b * T + t) * D;
    float acc = 0.0f;
    for (int d = 0; d < D; ++d) acc += sdo[d] * sp[d];
    sdot[t] = acc;
  }
  __syncthreads();
  float lw = 0.0f;
  for (int t = threadIdx.x; t < T; t += blockDim.x)
    lw += w[b * T + t] * sdot[t];
  red[threadIdx.x] = lw;
  __syncthreads();
  for (int off = blockDim.x >> 1; off > 0; off >>= 1) {
    if (threadIdx.x < off) red[threadIdx.x] += red[threadIdx.x + off];
    __syncthreads();
  }
  const float wdot = red[0];
  for (int t = threadIdx.x; t < T; t += blockDim.x) {
    const float wt = w[b * T + t];
    dscores[b * T + t] =
        mask[b * T + t] ? wt * (sdot[t] - wdot) : 0.0f;
  }
  for (int i = threadIdx.x; i < T * D; i += blockDim.x) {
    const int t = i / D;
    dseq[(int64_t)b * T * D + i] = w[b * T + t] * sdo[i - t * D];
  }
}

}  // namespace

static const short* att_bf_ptr(const torch::Tensor& t) {
  return reinterpret_cast<const short*>(t.data_ptr<at::BFloat16>());
}
static short* att_bf_ptr_mut(torch::Tensor& t) {
  return reinterpret_cast<short*>(t.data_ptr<at::BFloat16>());
}

std::tuple<torch::Tensor, torch::Tensor> mha_fwd(torch::Tensor q,
                                                 torch::Tensor k,
                                                 torch::Tensor v,
                                                 torch::Tensor key_pad,
                                                 int64_t n_heads,
                                                 double scale) {
  TORCH_CHECK(q.scalar_type() == torch::kBFloat16 && q.is_contiguous());
  TORCH_CHECK(key_pad.scalar_type() == torch::kUInt8);
  int B = q.size(0), T = q.size(1), D = q.size(2);
  int H = (int)n_heads;
  TORCH_CHECK(D % H == 0 && D / H <= kMaxDh, "head dim too large");
  TORCH_CHECK(T * H <= 1024, "sequence too long for the per-sample block");
  auto out = torch::empty_like(q);
  auto stats = torch::empty({B, (int64_t)H, (int64_t)T, 2},
                            q.options().dtype(torch::kFloat32));
  if (B == 0) return {out, stats};
  size_t lds = 2 * (size_t)T * D * sizeof(short);
  int threads = std::min(1024, ((T * H + 63) / 64) * 64);
  const int dh = D / H;
  TORCH_CHECK(dh == 4 || dh == 8 || dh == 16 || dh == 32,
              "head dim must be 4/8/16/32");
#define MHA_FWD(DHV) \
  k_mha_fwd<DHV><<<B, threads, lds, att_stream()>>>( \
      att_bf_ptr(q), att_bf_ptr(k), att_bf_ptr(v), \
      key_pad.data_ptr<uint8_t>(), B, T, D, H, (float)scale, \
      att_bf_ptr_mut(out), stats.data_ptr<float>())
  if (dh == 4) MHA_FWD(4);
  else if (dh == 8) MHA_FWD(8);
  else if (dh == 16) MHA_FWD(16);
  else MHA_FWD(32);
#undef MHA_FWD
  return {out, stats};
}

std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> mha_bwd(
    torch::Tensor dout, torch::Tensor q, torch::Tensor k, torch::Tensor v,
    torch::Tensor key_pad, torch::Tensor stats, int64_t n_heads,
    double scale) {
  int B = q.size(0), T = q.size(1), D = q.size(2);
  int H = (int)n_heads;
  auto dq = torch::empty_like(q);
  auto dk = torch::empty_like(k);
  auto dv = torch::empty_like(v);
  if (B == 0) return {dq, dk, dv};
  size_t lds = 4 * (size_t)T * D * sizeof(short)
               + (size_t)T * H * sizeof(float);
  int threads = std::min(1024, ((T * H + 63) / 64) * 64);
  const int dh = D / H;
#define MHA_BWD(DHV) \
  k_mha_bwd<DHV><<<B, threads, lds, att_stream()>>>( \
      att_bf_ptr(dout.contiguous()), att_bf_ptr(q), att_bf_ptr(k), \
      att_bf_ptr(v), key_pad.data_ptr<uint8_t>(), stats.data_ptr<float>(), \
      B, T, D, H, (float)scale, att_bf_ptr_mut(dq), att_bf_ptr_mut(dk), \
      att_bf_ptr_mut(dv))
  if (dh == 4) MHA_BWD(4);
  else if (dh == 8) MHA_BWD(8);
  else if (dh == 16) MHA_BWD(16);
  else MHA_BWD(32);
#undef MHA_BWD
  return {dq, dk, dv};
}

torch::Tensor din_feat_fwd(torch::Tensor seq, torch::Tensor tgt) {
  TORCH_CHECK(seq.is_contiguous() && seq.scalar_type() == torch::kFloat32);
  int B = seq.size(0), T = seq.size(1), D = seq.size(2);
  auto out = torch::empty({(int64_t)B * T, 4 * (int64_t)D},
                          seq.options().dtype(torch::kBFloat16));
  int64_t total = (int64_t)B * T * D;
  int blocks = (int)std::min<int64_t>((total + 255) / 256, 8192);
  k_din_feat_fwd<<<blocks, 256, 0, att_stream()>>>(
      seq.data_ptr<float>(), tgt.data_ptr<float>(), B, T, D,
      att_bf_ptr_mut(out));
  return out;
}

std::tuple<torch::Tensor, torch::Tensor> din_feat_bwd(torch::Tensor g,
                                                      torch::Tensor seq,
                                                      torch::Tensor tgt) {
  int B = seq.size(0), T = seq.size(1), D = seq.size(2);
  auto dseq = torch::empty_like(seq);
  auto dtgt = torch::empty_like(tgt);
  auto gc = g.contiguous();
  int64_t total = (int64_t)B * T * D;
  int blocks = (int)std::min<int64_t>((total + 255) / 256, 8192);
  k_din_feat_bwd_ds<<<blocks, 256, 0, att_stream()>>>(
      att_bf_ptr(gc), tgt.data_ptr<float>(), B, T, D,
      dseq.data_ptr<float>());
  int blocks2 = (int)std::min<int64_t>(((int64_t)B * D + 255) / 256, 8192);
  k_din_feat_bwd_dt<<<blocks2, 256, 0, att_stream()>>>(
      att_bf_ptr(gc), seq.data_ptr<float>(), B, T, D,
      dtgt.data_ptr<float>());
  return {dseq, dtgt};
}

std::tuple<torch::Tensor, torch::Tensor> msm_pool_fwd(torch::Tensor scores,
                                                      torch::Tensor seq,
                                                      torch::Tensor mask) {
  TORCH_CHECK(scores.scalar_type() == torch::kFloat32
                  && scores.is_contiguous());
  TORCH_CHECK(seq.scalar_type() == torch::kFloat32 && seq.is_contiguous());
  TORCH_CHECK(mask.scalar_type() == torch::kUInt8 && mask.is_contiguous());
  int B = scores.size(0), T = scores.size(1), D = seq.size(2);
  TORCH_CHECK(seq.size(0) == B && seq.size(1) == T && mask.numel() == B * T);
  TORCH_CHECK(T <= 4096, "sequence too long for the per-sample block");
  auto w = torch::empty_like(scores);
  auto out = torch::empty({(int64_t)B, (int64_t)D}, seq.options());
  if (B == 0) return {out, w};
  k_msm_pool_fwd<<<B, 128, (size_t)T * sizeof(float), att_stream()>>>(
      scores.data_ptr<float>(), seq.data_ptr<float>(),
      mask.data_ptr<uint8_t>(), B, T, D, w.data_ptr<float>(),
      out.data_ptr<float>());
  return {out, w};
}

std::tuple<torch::Tensor, torch::Tensor> msm_pool_bwd(torch::Tensor dout,
                                                      torch::Tensor seq,
                                                      torch::Tensor w,
                                                      torch::Tensor mask) {
  int B = seq.size(0), T = seq.size(1), D = seq.size(2);
  auto dc = dout.contiguous();
  auto dscores = torch::empty_like(w);
  auto dseq = torch::empty_like(seq);
  if (B == 0) return {dscores, dseq};
  k_msm_pool_bwd<<<B, 128, (size_t)(T + D) * sizeof(float), att_stream()>>>(
      dc.data_ptr<float>(), seq.data_ptr<float>(), w.data_ptr<float>(),
      mask.data_ptr<uint8_t>(), B, T, D, dscores.data_ptr<float>(),
      dseq.data_ptr<float>());
  return {dscores, dseq};
}

void register_attention(pybind11::module_& mod) {
  mod.def("mha_fwd", &mha_fwd);
  mod.def("mha_bwd", &mha_bwd);
  mod.def("din_feat_fwd", &din_feat_fwd);
  mod.def("din_feat_bwd", &din_feat_bwd);
  mod.def("msm_pool_fwd", &msm_pool_fwd);
  mod.def("msm_pool_bwd", &msm_pool_bwd);
}
