// Stochastic-rounding kernels: fp32->bf16 SR copy and fused AdamW step with
// SR bf16 parameter writes. Replaces the reference's Triton kernels
// (d9d/kernel/stochastic/copy.py, d9d/kernel/stochastic/adamw_step.py).
//
// RNG is counter-based (splitmix64 of seed ^ element index): stateless,
// reproducible given the host-provided seed, no curand/rocrand state.
#include "common.h"

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

namespace d9d {

template <int BLOCK>
__global__ void copy_fp32_to_bf16_sr_kernel(
    const float* __restrict__ src,
    ushort* __restrict__ dst,
    int64_t n, uint64_t seed) {
  const int64_t n_vec = n / 4;
  const int64_t tid = blockIdx.x * static_cast<int64_t>(BLOCK) + threadIdx.x;
  const int64_t stride = static_cast<int64_t>(gridDim.x) * BLOCK;
  for (int64_t v = tid; v < n_vec; v += stride) {
    const float4v in = *reinterpret_cast<const float4v*>(src + v * 4);
    const uint64_t r = splitmix64(seed ^ static_cast<uint64_t>(v));
    ushort4v out;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      out[j] = f32_to_bf16_stochastic(in[j], static_cast<uint32_t>(r >> (16 * j)));
    }
    *reinterpret_cast<ushort4v*>(dst + v * 4) = out;
  }
  for (int64_t i = n_vec * 4 + tid; i < n; i += stride) {
    const uint64_t r = splitmix64(seed ^ (0x8000000000000000ull | static_cast<uint64_t>(i)));
    dst[i] = f32_to_bf16_stochastic(src[i], static_cast<uint32_t>(r));
  }
}

// Fused AdamW: bf16 params+grads, fp32 moments, fp32 math, SR bf16 writes.
template <int BLOCK>
__global__ void adamw_sr_bf16_kernel(
    ushort* __restrict__ p,         // (n,) bf16 param
    const ushort* __restrict__ g,   // (n,) bf16 grad
    float* __restrict__ m,          // (n,) fp32 exp_avg
    float* __restrict__ v,          // (n,) fp32 exp_avg_sq
    int64_t n,
    float lr, float beta1, float beta2, float eps, float weight_decay,
    float bias_corr1, float bias_corr2,  // 1 - beta^t, precomputed on host
    uint64_t seed) {
  const int64_t n_vec = n / 4;
  const int64_t tid = blockIdx.x * static_cast<int64_t>(BLOCK) + threadIdx.x;
  const int64_t stride = static_cast<int64_t>(gridDim.x) * BLOCK;
  const float decay_mul = 1.f - lr * weight_decay;
  const float inv_bc1 = 1.f / bias_corr1;
  const float inv_sqrt_bc2 = rsqrtf(bias_corr2);

  for (int64_t vi = tid; vi < n_vec; vi += stride) {
    ushort4v pv = *reinterpret_cast<const ushort4v*>(p + vi * 4);
    const ushort4v gv = *reinterpret_cast<const ushort4v*>(g + vi * 4);
    float4v mv = *reinterpret_cast<const float4v*>(m + vi * 4);
    float4v vv = *reinterpret_cast<const float4v*>(v + vi * 4);
    const uint64_t r = splitmix64(seed ^ static_cast<uint64_t>(vi));
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float pf = bf16_bits_to_f32(pv[j]) * decay_mul;
      const float gf = bf16_bits_to_f32(gv[j]);
      float mf = beta1 * mv[j] + (1.f - beta1) * gf;
      float vf = beta2 * vv[j] + (1.f - beta2) * gf * gf;
      const float m_hat = mf * inv_bc1;
      const float denom = sqrtf(vf) * inv_sqrt_bc2 + eps;
      pf -= lr * m_hat / denom;
      mv[j] = mf;
      vv[j] = vf;
      pv[j] = f32_to_bf16_stochastic(pf, static_cast<uint32_t>(r >> (16 * j)));
    }
    *reinterpret_cast<ushort4v*>(p + vi * 4) = pv;
    *reinterpret_cast<float4v*>(m + vi * 4) = mv;
    *reinterpret_cast<float4v*>(v + vi * 4) = vv;
  }
  for (int64_t i = n_vec * 4 + tid; i < n; i += stride) {
    float pf = bf16_bits_to_f32(p[i]) * decay_mul;
    const float gf = bf16_bits_to_f32(g[i]);
    float mf = beta1 * m[i] + (1.f - beta1) * gf;
    float vf = beta2 * v[i] + (1.f - beta2) * gf * gf;
    pf -= lr * (mf * inv_bc1) / (sqrtf(vf) * inv_sqrt_bc2 + eps);
    m[i] = mf;
    v[i] = vf;
    const uint64_t r = splitmix64(seed ^ (0x8000000000000000ull | static_cast<uint64_t>(i)));
    p[i] = f32_to_bf16_stochastic(pf, static_cast<uint32_t>(r));
  }
}


// Multi-tensor fused AdamW: one launch for the whole parameter list (the
// per-tensor variant costs ~360 launches per optimizer step). Slots of 8
// elements, software-pipelined one slot ahead; each thread binary-searches
// its first tensor in the prefix table once. Per-element SR noise is
// bitwise identical to the single-tensor kernel (4-vector salt = elem/4).
template <int BLOCK>
__global__ void adamw_sr_bf16_multi_kernel(
    const int64_t* __restrict__ meta_g,  // prefix(N+1) | count(N) | p|g|m|v ptrs (N each) | seed(N)
    const float* __restrict__ bc_g,      // (N, 2): bias_corr1, bias_corr2
    int n_tensors, int64_t total_slots,
    float lr, float beta1, float beta2, float eps, float weight_decay) {
  // Stage the lookup tables in LDS: the per-slot binary search is a chain of
  // dependent loads, and from global memory it dominates the whole kernel.
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  int64_t* meta = reinterpret_cast<int64_t*>(smem_raw);
  float* bc = reinterpret_cast<float*>(meta + 7 * n_tensors + 1);
  for (int i = threadIdx.x; i < 7 * n_tensors + 1; i += BLOCK) meta[i] = meta_g[i];
  for (int i = threadIdx.x; i < 2 * n_tensors; i += BLOCK) bc[i] = bc_g[i];
  __syncthreads();
  const int64_t* prefix = meta;
  const int64_t* count = meta + n_tensors + 1;
  const int64_t* p_ptrs = count + n_tensors;
  const int64_t* g_ptrs = p_ptrs + n_tensors;
  const int64_t* m_ptrs = g_ptrs + n_tensors;
  const int64_t* v_ptrs = m_ptrs + n_tensors;
  const int64_t* seeds = v_ptrs + n_tensors;
  const float decay_mul = 1.f - lr * weight_decay;

  const int64_t tid = blockIdx.x * static_cast<int64_t>(BLOCK) + threadIdx.x;
  const int64_t stride = static_cast<int64_t>(gridDim.x) * BLOCK;
  // ONE binary search to place the thread's first slot; the grid-stride
  // walk advances t monotonically after that (a 9-step dependent-load
  // search per slot was ~2x the whole kernel's useful time at 1.6e9
  // parameters).
  int t = 0;
  {
    int lo = 0, hi = n_tensors - 1;
    while (lo < hi) {
      const int mid = (lo + hi + 1) >> 1;
      if (prefix[mid] <= min(tid, total_slots - 1)) lo = mid; else hi = mid - 1;
    }
    t = lo;
  }

  // 8-element slots, software-pipelined one slot ahead: vmcnt completion is
  // IN-ORDER on CDNA, so a store-then-load iteration shape makes every
  // iteration's load wait drain the previous stores too. Issuing the next
  // slot's loads BEFORE the current slot's stores breaks that chain (the
  // 4-element store-then-load version measured ~1.9 TB/s effective).
  struct SlotData {
    ushort4v pv0, pv1, gv0, gv1;
    float4v mv0, mv1, vv0, vv1;
    int64_t base, n;
    int t;
    bool full;
  };
  auto load_slot = [&](int64_t slot, int& tt, SlotData& s) {
    while (tt + 1 < n_tensors && prefix[tt + 1] <= slot) ++tt;
    const int64_t vi = slot - prefix[tt];
    s.t = tt;
    s.n = count[tt];
    s.base = vi * 8;
    s.full = s.base + 8 <= s.n;
    if (s.full) {
      const ushort* p = reinterpret_cast<const ushort*>(p_ptrs[tt]);
      const ushort* g = reinterpret_cast<const ushort*>(g_ptrs[tt]);
      const float* m = reinterpret_cast<const float*>(m_ptrs[tt]);
      const float* v = reinterpret_cast<const float*>(v_ptrs[tt]);
      s.pv0 = *reinterpret_cast<const ushort4v*>(p + s.base);
      s.pv1 = *reinterpret_cast<const ushort4v*>(p + s.base + 4);
      s.gv0 = *reinterpret_cast<const ushort4v*>(g + s.base);
      s.gv1 = *reinterpret_cast<const ushort4v*>(g + s.base + 4);
      s.mv0 = *reinterpret_cast<const float4v*>(m + s.base);
      s.mv1 = *reinterpret_cast<const float4v*>(m + s.base + 4);
      s.vv0 = *reinterpret_cast<const float4v*>(v + s.base);
      s.vv1 = *reinterpret_cast<const float4v*>(v + s.base + 4);
    }
  };
  auto compute_store = [&](SlotData& s) {
    const float inv_bc1 = 1.f / bc[s.t * 2];
    const float inv_sqrt_bc2 = rsqrtf(bc[s.t * 2 + 1]);
    if (s.full) {
      const int64_t vi = s.base >> 3;
      const uint64_t r0 = splitmix64((uint64_t)seeds[s.t] ^ (uint64_t)(vi << 1));
      const uint64_t r1 =
          splitmix64((uint64_t)seeds[s.t] ^ (uint64_t)((vi << 1) | 1));
      ushort4v* pvs[2] = {&s.pv0, &s.pv1};
      const ushort4v* gvs[2] = {&s.gv0, &s.gv1};
      float4v* mvs[2] = {&s.mv0, &s.mv1};
      float4v* vvs[2] = {&s.vv0, &s.vv1};
      const uint64_t rs[2] = {r0, r1};
#pragma unroll
      for (int h = 0; h < 2; ++h) {
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          float pf = bf16_bits_to_f32((*pvs[h])[j]) * decay_mul;
          const float gf = bf16_bits_to_f32((*gvs[h])[j]);
          float mf = beta1 * (*mvs[h])[j] + (1.f - beta1) * gf;
          float vf = beta2 * (*vvs[h])[j] + (1.f - beta2) * gf * gf;
          pf -= lr * (mf * inv_bc1) / (sqrtf(vf) * inv_sqrt_bc2 + eps);
          (*mvs[h])[j] = mf;
          (*vvs[h])[j] = vf;
          (*pvs[h])[j] =
              f32_to_bf16_stochastic(pf, static_cast<uint32_t>(rs[h] >> (16 * j)));
        }
      }
      ushort* p = reinterpret_cast<ushort*>(p_ptrs[s.t]);
      float* m = reinterpret_cast<float*>(m_ptrs[s.t]);
      float* v = reinterpret_cast<float*>(v_ptrs[s.t]);
      *reinterpret_cast<ushort4v*>(p + s.base) = s.pv0;
      *reinterpret_cast<ushort4v*>(p + s.base + 4) = s.pv1;
      *reinterpret_cast<float4v*>(m + s.base) = s.mv0;
      *reinterpret_cast<float4v*>(m + s.base + 4) = s.mv1;
      *reinterpret_cast<float4v*>(v + s.base) = s.vv0;
      *reinterpret_cast<float4v*>(v + s.base + 4) = s.vv1;
    } else {
      // tail slot, bitwise-matching the single-tensor kernel: any full
      // 4-group left in the window uses the 4-vector RNG (salt = i/4),
      // the rest the per-element RNG.
      ushort* p = reinterpret_cast<ushort*>(p_ptrs[s.t]);
      const ushort* g = reinterpret_cast<const ushort*>(g_ptrs[s.t]);
      float* m = reinterpret_cast<float*>(m_ptrs[s.t]);
      float* v = reinterpret_cast<float*>(v_ptrs[s.t]);
      int64_t i4 = s.base;
      for (; i4 + 4 <= s.n; i4 += 4) {
        ushort4v pv = *reinterpret_cast<const ushort4v*>(p + i4);
        const ushort4v gv = *reinterpret_cast<const ushort4v*>(g + i4);
        float4v mv = *reinterpret_cast<const float4v*>(m + i4);
        float4v vv = *reinterpret_cast<const float4v*>(v + i4);
        const uint64_t r =
            splitmix64((uint64_t)seeds[s.t] ^ (uint64_t)(i4 >> 2));
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          float pf = bf16_bits_to_f32(pv[j]) * decay_mul;
          const float gf = bf16_bits_to_f32(gv[j]);
          float mf = beta1 * mv[j] + (1.f - beta1) * gf;
          float vf = beta2 * vv[j] + (1.f - beta2) * gf * gf;
          pf -= lr * (mf * inv_bc1) / (sqrtf(vf) * inv_sqrt_bc2 + eps);
          mv[j] = mf;
          vv[j] = vf;
          pv[j] = f32_to_bf16_stochastic(pf, static_cast<uint32_t>(r >> (16 * j)));
        }
        *reinterpret_cast<ushort4v*>(p + i4) = pv;
        *reinterpret_cast<float4v*>(m + i4) = mv;
        *reinterpret_cast<float4v*>(v + i4) = vv;
      }
      for (int64_t i = i4; i < s.n; ++i) {
        float pf = bf16_bits_to_f32(p[i]) * decay_mul;
        const float gf = bf16_bits_to_f32(g[i]);
        float mf = beta1 * m[i] + (1.f - beta1) * gf;
        float vf = beta2 * v[i] + (1.f - beta2) * gf * gf;
        pf -= lr * (mf * inv_bc1) / (sqrtf(vf) * inv_sqrt_bc2 + eps);
        m[i] = mf;
        v[i] = vf;
        const uint64_t rt = splitmix64(
            (uint64_t)seeds[s.t] ^ (0x8000000000000000ull | (uint64_t)i));
        p[i] = f32_to_bf16_stochastic(pf, static_cast<uint32_t>(rt));
      }
    }
  };

  int64_t slot = tid;
  if (slot >= total_slots) return;
  SlotData cur;
  load_slot(slot, t, cur);
  while (true) {
    const int64_t nslot = slot + stride;
    if (nslot >= total_slots) {
      compute_store(cur);
      break;
    }
    SlotData nxt;
    load_slot(nslot, t, nxt);  // next loads issue BEFORE cur's stores
    compute_store(cur);
    cur = nxt;
    slot = nslot;
  }
}

}  // namespace d9d

namespace d9d {

// acc (fp32) += x (bf16), one pass: the CCE backward accumulates the
// classifier gradient over row chunks and the torch chain
// (part.float(); dc += part32) costs an extra fp32 round-trip per chunk
// (~0.23 ms x chunks per backward at V=152k).
template <int BLOCK>
__global__ void add_bf16_into_f32_kernel(
    float* __restrict__ acc, const ushort* __restrict__ x, int64_t n) {
  const int64_t stride = (int64_t)gridDim.x * BLOCK * 8;
  for (int64_t i = ((int64_t)blockIdx.x * BLOCK + threadIdx.x) * 8; i + 8 <= n;
       i += stride) {
    ushort4v x0 = *reinterpret_cast<const ushort4v*>(x + i);
    ushort4v x1 = *reinterpret_cast<const ushort4v*>(x + i + 4);
    float4v a0 = *reinterpret_cast<const float4v*>(acc + i);
    float4v a1 = *reinterpret_cast<const float4v*>(acc + i + 4);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      a0[j] += bf16_bits_to_f32(x0[j]);
      a1[j] += bf16_bits_to_f32(x1[j]);
    }
    *reinterpret_cast<float4v*>(acc + i) = a0;
    *reinterpret_cast<float4v*>(acc + i + 4) = a1;
  }
  // tail
  const int64_t t0 = (n / 8) * 8;
  for (int64_t i = t0 + blockIdx.x * BLOCK + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * BLOCK) {
    acc[i] += bf16_bits_to_f32(x[i]);
  }
}

}  // namespace d9d

void add_bf16_into_f32_(torch::Tensor acc, torch::Tensor x) {
  TORCH_CHECK(acc.is_cuda() && acc.scalar_type() == torch::kFloat32 &&
              acc.is_contiguous());
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16 && x.is_contiguous());
  TORCH_CHECK(acc.numel() == x.numel());
  const int64_t n = acc.numel();
  if (n == 0) return;
  constexpr int kBlock = 256;
  const int grid = (int)std::min<int64_t>((n / 8 + kBlock - 1) / kBlock, 16384);
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL((d9d::add_bf16_into_f32_kernel<kBlock>),
                     dim3(std::max(grid, 1)), dim3(kBlock), 0, stream,
                     acc.data_ptr<float>(),
                     reinterpret_cast<const ushort*>(x.data_ptr()), n);
}

void copy_fp32_to_bf16_stochastic_(
    torch::Tensor dst, torch::Tensor src, int64_t seed) {
  TORCH_CHECK(src.is_cuda() && src.scalar_type() == torch::kFloat32 && src.is_contiguous());
  TORCH_CHECK(dst.scalar_type() == torch::kBFloat16 && dst.is_contiguous());
  TORCH_CHECK(dst.numel() == src.numel());
  const int64_t n = src.numel();
  if (n == 0) return;
  constexpr int kBlock = 256;
  const int grid = static_cast<int>(
      std::min<int64_t>(d9d::ceil_div(n / 4 + 1, kBlock), 2048));
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(
      (d9d::copy_fp32_to_bf16_sr_kernel<kBlock>), dim3(grid), dim3(kBlock), 0,
      stream, src.data_ptr<float>(),
      reinterpret_cast<ushort*>(dst.data_ptr()), n,
      static_cast<uint64_t>(seed));
}

void adamw_stochastic_bf16_(
    torch::Tensor p, torch::Tensor g, torch::Tensor m, torch::Tensor v,
    double lr, double beta1, double beta2, double eps, double weight_decay,
    int64_t step, int64_t seed) {
  TORCH_CHECK(p.is_cuda() && p.scalar_type() == torch::kBFloat16 && p.is_contiguous());
  TORCH_CHECK(g.scalar_type() == torch::kBFloat16 && g.is_contiguous());
  TORCH_CHECK(m.scalar_type() == torch::kFloat32 && m.is_contiguous());
  TORCH_CHECK(v.scalar_type() == torch::kFloat32 && v.is_contiguous());
  const int64_t n = p.numel();
  TORCH_CHECK(g.numel() == n && m.numel() == n && v.numel() == n);
  if (n == 0) return;

  const float bc1 = 1.f - powf(static_cast<float>(beta1), static_cast<float>(step));
  const float bc2 = 1.f - powf(static_cast<float>(beta2), static_cast<float>(step));

  constexpr int kBlock = 256;
  const int grid = static_cast<int>(
      std::min<int64_t>(d9d::ceil_div(n / 4 + 1, kBlock), 2048));
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(
      (d9d::adamw_sr_bf16_kernel<kBlock>), dim3(grid), dim3(kBlock), 0, stream,
      reinterpret_cast<ushort*>(p.data_ptr()),
      reinterpret_cast<const ushort*>(g.data_ptr()),
      m.data_ptr<float>(), v.data_ptr<float>(), n,
      static_cast<float>(lr), static_cast<float>(beta1),
      static_cast<float>(beta2), static_cast<float>(eps),
      static_cast<float>(weight_decay), bc1, bc2, static_cast<uint64_t>(seed));
}


void adamw_stochastic_bf16_multi_(
    std::vector<torch::Tensor> params, std::vector<torch::Tensor> grads,
    std::vector<torch::Tensor> exp_avgs, std::vector<torch::Tensor> exp_avg_sqs,
    double lr, double beta1, double beta2, double eps, double weight_decay,
    std::vector<int64_t> steps, std::vector<int64_t> seeds) {
  const int n = (int)params.size();
  if (n == 0) return;
  auto meta_cpu = torch::empty({(int64_t)n * 7 + 1}, torch::dtype(torch::kInt64));
  auto bc_cpu = torch::empty({(int64_t)n * 2}, torch::dtype(torch::kFloat32));
  int64_t* mp = meta_cpu.data_ptr<int64_t>();
  float* bp = bc_cpu.data_ptr<float>();
  int64_t slots = 0;
  for (int i = 0; i < n; ++i) {
    TORCH_CHECK(params[i].is_cuda() && params[i].scalar_type() == torch::kBFloat16);
    TORCH_CHECK(grads[i].is_contiguous() && params[i].is_contiguous());
    const int64_t cnt = params[i].numel();
    mp[i] = slots;                          // prefix
    mp[n + 1 + i] = cnt;                    // count
    mp[2 * n + 1 + i] = (int64_t)params[i].data_ptr();
    mp[3 * n + 1 + i] = (int64_t)grads[i].data_ptr();
    mp[4 * n + 1 + i] = (int64_t)exp_avgs[i].data_ptr();
    mp[5 * n + 1 + i] = (int64_t)exp_avg_sqs[i].data_ptr();
    mp[6 * n + 1 + i] = seeds[i];
    bp[i * 2] = 1.f - powf((float)beta1, (float)steps[i]);
    bp[i * 2 + 1] = 1.f - powf((float)beta2, (float)steps[i]);
    slots += (cnt + 7) / 8;
  }
  mp[n] = slots;
  auto meta = meta_cpu.to(params[0].device(), /*non_blocking=*/true);
  auto bc = bc_cpu.to(params[0].device(), /*non_blocking=*/true);
  constexpr int kBlock = 256;
  const int grid = (int)std::min<int64_t>((slots + kBlock - 1) / kBlock, 16384);
  const size_t smem = (7 * (size_t)n + 1) * 8 + 2 * (size_t)n * 4;
  TORCH_CHECK(smem <= 64 * 1024, "adamw multi: too many tensors for LDS");
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL((d9d::adamw_sr_bf16_multi_kernel<kBlock>), dim3(grid),
                     dim3(kBlock), smem, stream, meta.data_ptr<int64_t>(),
                     bc.data_ptr<float>(), n, slots, (float)lr, (float)beta1,
                     (float)beta2, (float)eps, (float)weight_decay);
}
