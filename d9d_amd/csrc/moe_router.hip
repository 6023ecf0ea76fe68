// Fused MoE top-k router for CDNA4.
//
// Replaces the torch chain softmax -> (+bias) -> topk -> gather -> renorm
// (reference: d9d/module/block/moe/router.py forward) with one wave-per-row
// kernel each way. The torch chain costs ~5 kernel launches per layer and
// the radix top-k alone is ~0.2 ms at (32768, 128); this does the whole row
// in registers (E <= 1024, k <= 16).
//
// Selection uses probs + expert_bias (aux-free load balancing) while the
// RETURNED probabilities are bias-free, renormalized over the selected k —
// identical semantics to the torch path, ties broken toward the lower
// expert index like torch.topk.
#include "common.h"

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

namespace d9d {

constexpr int kMaxCols = 16;  // E <= 16*64
constexpr int kMaxK = 16;

// Sortable u32 from float (IEEE): order-preserving bijection.
D9D_DEVICE uint32_t f32_sortable(float x) {
  uint32_t u = __builtin_bit_cast(uint32_t, x);
  return (u & 0x80000000u) ? ~u : (u | 0x80000000u);
}

template <int BLOCK>
__global__ void router_topk_fwd_kernel(
    const float* __restrict__ logits,  // (T, E)
    const float* __restrict__ bias,    // (E,) or nullptr; selection only
    float* __restrict__ top_probs,     // (T, K)
    int64_t* __restrict__ top_idx,     // (T, K)
    int64_t T, int E, int K, int renorm) {
  const int lane = threadIdx.x & 63;
  const int64_t row = (int64_t)blockIdx.x * (BLOCK / 64) + (threadIdx.x >> 6);
  if (row >= T) return;
  const int nv = (E + 63) / 64;

  float v[kMaxCols];
#pragma unroll
  for (int i = 0; i < kMaxCols; ++i) {
    if (i >= nv) break;
    const int c = i * 64 + lane;
    v[i] = (c < E) ? logits[row * E + c] : -1e30f;
  }
  // softmax over the row (fp32)
  float m = -1e30f;
#pragma unroll
  for (int i = 0; i < kMaxCols; ++i) {
    if (i >= nv) break;
    m = fmaxf(m, v[i]);
  }
#pragma unroll
  for (int off = 1; off < 64; off <<= 1) m = fmaxf(m, __shfl_xor(m, off, 64));
  float sum = 0.f;
#pragma unroll
  for (int i = 0; i < kMaxCols; ++i) {
    if (i >= nv) break;
    v[i] = __builtin_amdgcn_exp2f((v[i] - m) * 1.44269504089f);
    sum += v[i];
  }
#pragma unroll
  for (int off = 1; off < 64; off <<= 1) sum += __shfl_xor(sum, off, 64);
  const float inv = 1.f / sum;
  float s[kMaxCols];  // selection scores (probs + bias)
#pragma unroll
  for (int i = 0; i < kMaxCols; ++i) {
    if (i >= nv) break;
    const int c = i * 64 + lane;
    v[i] *= inv;  // v is now the probability
    s[i] = (c < E) ? (bias ? v[i] + bias[c] : v[i]) : -1e30f;
  }

  // iterative top-K: wave-argmax on a sortable (score, ~idx) key
  float tp[kMaxK];
  int ti[kMaxK];
  float ssum = 0.f;
  for (int k = 0; k < K; ++k) {
    uint64_t key = 0;
#pragma unroll
    for (int i = 0; i < kMaxCols; ++i) {
      if (i >= nv) break;
      const int c = i * 64 + lane;
      const uint64_t cand =
          ((uint64_t)f32_sortable(s[i]) << 32) | (uint32_t)(0x7fffffff - c);
      key = (cand > key) ? cand : key;
    }
#pragma unroll
    for (int off = 1; off < 64; off <<= 1) {
      const uint64_t o = __shfl_xor((long long)key, off, 64);
      key = (o > key) ? o : key;
    }
    const int idx = 0x7fffffff - (int)(key & 0xffffffffu);
    // broadcast the winner's probability; owner masks it for the next pass
    float cand_p = 0.f;
#pragma unroll
    for (int i = 0; i < kMaxCols; ++i) {
      if (i >= nv) break;
      if (i == (idx >> 6)) {
        cand_p = v[i];
        if (lane == (idx & 63)) s[i] = -1e30f;
      }
    }
    tp[k] = __shfl(cand_p, idx & 63, 64);
    ti[k] = idx;
    ssum += tp[k];
  }

  // constant-index writes: tp/ti are identical on every lane (shfl
  // broadcast), so lane 0 writes them all -- a tp[lane] dynamic index would
  // push the whole array to scratch.
  const float norm = renorm ? 1.f / fmaxf(ssum, 1e-20f) : 1.f;
  if (lane == 0) {
#pragma unroll
    for (int k = 0; k < kMaxK; ++k) {
      if (k >= K) break;
      top_probs[row * K + k] = tp[k] * norm;
      top_idx[row * K + k] = ti[k];
    }
  }
}

// dlogits from dtop_probs: renorm chain then softmax backward, recomputing
// the softmax from the saved logits (cheaper than materializing (T,E) probs).
template <int BLOCK>
__global__ void router_topk_bwd_kernel(
    const float* __restrict__ logits,    // (T, E)
    const int64_t* __restrict__ top_idx, // (T, K)
    const float* __restrict__ dtop,      // (T, K)
    float* __restrict__ dlogits,         // (T, E)
    int64_t T, int E, int K, int renorm) {
  const int lane = threadIdx.x & 63;
  const int64_t row = (int64_t)blockIdx.x * (BLOCK / 64) + (threadIdx.x >> 6);
  if (row >= T) return;
  const int nv = (E + 63) / 64;

  float v[kMaxCols];
  float m = -1e30f;
#pragma unroll
  for (int i = 0; i < kMaxCols; ++i) {
    if (i >= nv) break;
    const int c = i * 64 + lane;
    v[i] = (c < E) ? logits[row * E + c] : -1e30f;
    m = fmaxf(m, v[i]);
  }
#pragma unroll
  for (int off = 1; off < 64; off <<= 1) m = fmaxf(m, __shfl_xor(m, off, 64));
  float sum = 0.f;
#pragma unroll
  for (int i = 0; i < kMaxCols; ++i) {
    if (i >= nv) break;
    v[i] = __builtin_amdgcn_exp2f((v[i] - m) * 1.44269504089f);
    sum += v[i];
  }
#pragma unroll
  for (int off = 1; off < 64; off <<= 1) sum += __shfl_xor(sum, off, 64);
  const float inv = 1.f / sum;
#pragma unroll
  for (int i = 0; i < kMaxCols; ++i) {
    if (i >= nv) break;
    v[i] *= inv;
  }

  // Selected-entry grads. S = sum p_sel; t_k = p_k / S;
  // renorm:  dp_k = (dtop_k - A) / S with A = sum_m dtop_m * t_m
  // no-norm: dp_k = dtop_k
  // Uniform loop over the k selected entries: ds_bpermute under divergence
  // returns undefined data from inactive source lanes, so every shfl here
  // runs with the full wave active. All lanes accumulate identical S/A.
  float S = 0.f, A = 0.f;
  for (int k = 0; k < K; ++k) {
    const int idx = (int)top_idx[row * K + k];
    float cand = 0.f;
#pragma unroll
    for (int i = 0; i < kMaxCols; ++i) {
      if (i >= nv) break;
      if (i == (idx >> 6)) cand = v[i];
    }
    const float p_k = __shfl(cand, idx & 63, 64);
    S += p_k;
    A += dtop[row * K + k] * p_k;
  }
  A /= fmaxf(S, 1e-20f);  // now A = sum dtop_m t_m

  // dp for selected entries, then softmax bwd: dz_j = p_j * (dp_j - B),
  // B = sum_j p_j dp_j (only selected entries have dp != 0)
  float dp[kMaxCols];
#pragma unroll
  for (int i = 0; i < kMaxCols; ++i) dp[i] = 0.f;
  float B = 0.f;
  for (int k = 0; k < K; ++k) {
    const int idx = (int)top_idx[row * K + k];
    const float dtk = dtop[row * K + k];
    const float dpk = renorm ? (dtk - A) / fmaxf(S, 1e-20f) : dtk;
#pragma unroll
    for (int i = 0; i < kMaxCols; ++i) {
      if (i >= nv) break;
      if (i == (idx >> 6) && lane == (idx & 63)) {
        dp[i] = dpk;
        B += v[i] * dpk;
      }
    }
  }
#pragma unroll
  for (int off = 1; off < 64; off <<= 1) B += __shfl_xor(B, off, 64);
#pragma unroll
  for (int i = 0; i < kMaxCols; ++i) {
    if (i >= nv) break;
    const int c = i * 64 + lane;
    if (c < E) dlogits[row * E + c] = v[i] * (dp[i] - B);
  }
}

}  // namespace d9d

std::vector<torch::Tensor> router_topk_fwd(
    torch::Tensor logits, c10::optional<torch::Tensor> bias, int64_t K,
    bool renormalize) {
  TORCH_CHECK(logits.is_cuda() && logits.scalar_type() == torch::kFloat32 &&
              logits.is_contiguous());
  const int64_t T = logits.size(0);
  const int E = logits.size(1);
  TORCH_CHECK(E <= 1024 && K <= 16, "router kernel supports E<=1024, k<=16");
  auto top_probs = torch::empty({T, K}, logits.options());
  auto top_idx = torch::empty({T, K}, logits.options().dtype(torch::kInt64));
  if (T == 0) return {top_probs, top_idx};
  const float* bias_ptr = nullptr;
  torch::Tensor bias_f;
  if (bias.has_value()) {
    bias_f = bias->to(torch::kFloat32).contiguous();
    bias_ptr = bias_f.data_ptr<float>();
  }
  constexpr int kBlock = 256;
  const int64_t grid = (T + (kBlock / 64) - 1) / (kBlock / 64);
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL((d9d::router_topk_fwd_kernel<kBlock>), dim3(grid),
                     dim3(kBlock), 0, stream, logits.data_ptr<float>(),
                     bias_ptr, top_probs.data_ptr<float>(),
                     top_idx.data_ptr<int64_t>(), T, E, (int)K,
                     renormalize ? 1 : 0);
  return {top_probs, top_idx};
}

torch::Tensor router_topk_bwd(
    torch::Tensor logits, torch::Tensor top_idx, torch::Tensor dtop,
    bool renormalize) {
  const int64_t T = logits.size(0);
  const int E = logits.size(1);
  const int K = top_idx.size(1);
  auto dlogits = torch::empty_like(logits);
  if (T == 0) return dlogits;
  constexpr int kBlock = 256;
  const int64_t grid = (T + (kBlock / 64) - 1) / (kBlock / 64);
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL((d9d::router_topk_bwd_kernel<kBlock>), dim3(grid),
                     dim3(kBlock), 0, stream, logits.data_ptr<float>(),
                     top_idx.data_ptr<int64_t>(),
                     dtop.contiguous().data_ptr<float>(),
                     dlogits.data_ptr<float>(), T, E, K, renormalize ? 1 : 0);
  return dlogits;
}
