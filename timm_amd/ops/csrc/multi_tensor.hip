// Multi-tensor fused optimizer kernels for gfx950.
// One launch updates every parameter tensor of the model (replaces the
// reference's torch._foreach_* loops: timm/optim/adamw.py:180,
// timm/utils/model_ema.py:227-231, utils/clip_grad.py:6).
//
// Tensor lists are flattened into a device-side pointer table + exclusive
// prefix-sum of numels; each thread binary-searches its tensor. HBM-bound.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace {

constexpr int kBlock = 256;

struct TensorTable {
  at::Tensor ptrs;     // int64 [n_lists, n_tensors] device
  at::Tensor prefix;   // int64 [n_tensors + 1] device
  long total;
  int n;
};

TensorTable build_table(const std::vector<std::vector<at::Tensor>>& lists) {
  int n = lists[0].size();
  int n_lists = lists.size();
  auto ptrs_cpu = at::empty({n_lists, n}, at::TensorOptions().dtype(at::kLong));
  auto prefix_cpu = at::empty({n + 1}, at::TensorOptions().dtype(at::kLong));
  long* pp = ptrs_cpu.data_ptr<long>();
  long* px = prefix_cpu.data_ptr<long>();
  long total = 0;
  for (int i = 0; i < n; ++i) {
    px[i] = total;
    total += lists[0][i].numel();
    for (int l = 0; l < n_lists; ++l) {
      pp[l * n + i] = (long)lists[l][i].data_ptr();
    }
  }
  px[n] = total;
  auto dev = lists[0][0].device();
  TensorTable t;
  t.ptrs = ptrs_cpu.to(dev, /*non_blocking=*/true);
  t.prefix = prefix_cpu.to(dev, /*non_blocking=*/true);
  t.total = total;
  t.n = n;
  return t;
}

__device__ __forceinline__ int find_tensor(const long* prefix, int n, long idx) {
  int lo = 0, hi = n;
  while (lo + 1 < hi) {
    int mid = (lo + hi) >> 1;
    if (prefix[mid] <= idx) lo = mid; else hi = mid;
  }
  return lo;
}

// ---- AdamW ----
// p in {fp32, bf16}, m/v fp32, g same dtype as p.

template <typename T>
__global__ void adamw_kernel(
    const long* __restrict__ ptrs, const long* __restrict__ prefix, int n, long total,
    float lr, float beta1, float beta2, float eps, float wd, float bc1, float bc2) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  const float step_size = lr / bc1;
  const float inv_bc2_sqrt = rsqrtf(bc2);
  const float decay = 1.f - lr * wd;
  for (; i < total; i += stride) {
    int t = find_tensor(prefix, n, i);
    long off = i - prefix[t];
    T* p = reinterpret_cast<T*>(ptrs[0 * n + t]);
    const T* g = reinterpret_cast<const T*>(ptrs[1 * n + t]);
    float* m = reinterpret_cast<float*>(ptrs[2 * n + t]);
    float* v = reinterpret_cast<float*>(ptrs[3 * n + t]);

    float gf = Elem<T>::to_f32(g[off]);
    float pf = Elem<T>::to_f32(p[off]) * decay;
    float mf = m[off] + (1.f - beta1) * (gf - m[off]);
    float vf = v[off] * beta2 + (1.f - beta2) * gf * gf;
    m[off] = mf;
    v[off] = vf;
    float denom = sqrtf(vf) * inv_bc2_sqrt + eps;
    pf -= step_size * mf / denom;
    p[off] = Elem<T>::from_f32(pf);
  }
}

// ---- lerp (EMA) ----  dst += w * (src - dst)

template <typename T>
__global__ void lerp_kernel(
    const long* __restrict__ ptrs, const long* __restrict__ prefix, int n, long total, float w) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (; i < total; i += stride) {
    int t = find_tensor(prefix, n, i);
    long off = i - prefix[t];
    T* d = reinterpret_cast<T*>(ptrs[0 * n + t]);
    const T* s = reinterpret_cast<const T*>(ptrs[1 * n + t]);
    float df = Elem<T>::to_f32(d[off]);
    float sf = Elem<T>::to_f32(s[off]);
    d[off] = Elem<T>::from_f32(df + w * (sf - df));
  }
}

// ---- global L2 norm ----

template <typename T>
__global__ void l2norm_kernel(
    const long* __restrict__ ptrs, const long* __restrict__ prefix, int n, long total,
    float* __restrict__ out) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  float acc = 0.f;
  for (; i < total; i += stride) {
    int t = find_tensor(prefix, n, i);
    long off = i - prefix[t];
    const T* p = reinterpret_cast<const T*>(ptrs[0 * n + t]);
    float f = Elem<T>::to_f32(p[off]);
    acc += f * f;
  }
  acc = wave_reduce_sum(acc);
  __shared__ float warp_sums[kBlock / WAVE_SIZE];
  int wave = threadIdx.x / WAVE_SIZE;
  int lane = threadIdx.x % WAVE_SIZE;
  if (lane == 0) warp_sums[wave] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    float s = 0.f;
    for (int w = 0; w < kBlock / WAVE_SIZE; ++w) s += warp_sums[w];
    atomicAdd(out, s);
  }
}


// ---- LAMB ----
// Two phases (per param group):
//  1. update m/v, form the raw Adam update u (+ attached weight decay),
//     OVERWRITING the grad buffer with u, while accumulating per-tensor
//     sum(p^2) and sum(u^2) (wave-segmented atomics, ~1 atomic/wave).
//  2. per-tensor trust ratio ||p|| / ||u|| (zero-guarded, optional clip at 1)
//     scales the step: p -= lr * trust * u.

template <typename T>
__global__ void lamb_phase1_kernel(
    const long* __restrict__ ptrs, const long* __restrict__ prefix, int n, long total,
    float beta1, float beta2, float beta3, float eps, float wd,
    float bc1, float bc2, float inv_clip,
    float* __restrict__ sq_acc) {            // [n, 2]: sum p^2, sum u^2
  const long stride = (long)gridDim.x * blockDim.x;
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const float inv_bc2_sqrt = rsqrtf(bc2);
  // converged-wave loop: every lane of a wave takes the same number of
  // iterations (inactive tail lanes contribute zeros), so the wave-level
  // reductions below stay well-defined
  const long wave_first = (long)blockIdx.x * blockDim.x + (threadIdx.x - lane);
  for (long base = wave_first; base < total; base += stride) {
    const long i = base + lane;
    const bool active = i < total;
    const long ci = active ? i : total - 1;
    const int t = find_tensor(prefix, n, ci);

    float pf = 0.f, uf = 0.f;
    if (active) {
      const long off = i - prefix[t];
      const T* p = reinterpret_cast<const T*>(ptrs[0 * n + t]);
      T* g = reinterpret_cast<T*>(ptrs[1 * n + t]);
      float* m = reinterpret_cast<float*>(ptrs[2 * n + t]);
      float* v = reinterpret_cast<float*>(ptrs[3 * n + t]);

      const float gf = Elem<T>::to_f32(g[off]) * inv_clip;
      pf = Elem<T>::to_f32(p[off]);
      const float mf = m[off] * beta1 + beta3 * gf;
      const float vf = v[off] * beta2 + (1.f - beta2) * gf * gf;
      m[off] = mf;
      v[off] = vf;
      const float denom = sqrtf(vf) * inv_bc2_sqrt + eps;
      uf = (mf / bc1) / denom + wd * pf;
      g[off] = Elem<T>::from_f32(uf);
    }

    // per-tensor squared-sum accumulation: a wave almost always lies inside
    // one tensor -> one atomic per wave; mixed waves fall back to per-lane
    const int t0 = __shfl(t, 0, 64);
    const bool uniform = __all(!active || t == t0);
    if (uniform) {
      float p2 = wave_reduce_sum(pf * pf);
      float u2 = wave_reduce_sum(uf * uf);
      if (lane == 0) {
        atomicAdd(&sq_acc[2 * t0 + 0], p2);
        atomicAdd(&sq_acc[2 * t0 + 1], u2);
      }
    } else if (active) {
      atomicAdd(&sq_acc[2 * t + 0], pf * pf);
      atomicAdd(&sq_acc[2 * t + 1], uf * uf);
    }
  }
}

template <typename T>
__global__ void lamb_phase2_kernel(
    const long* __restrict__ ptrs, const long* __restrict__ prefix, int n, long total,
    float lr, int adapt, int trust_clip,
    const float* __restrict__ sq_acc) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (; i < total; i += stride) {
    int t = find_tensor(prefix, n, i);
    long off = i - prefix[t];
    T* p = reinterpret_cast<T*>(ptrs[0 * n + t]);
    const T* g = reinterpret_cast<const T*>(ptrs[1 * n + t]);
    float trust = 1.f;
    if (adapt) {
      float w_norm = sqrtf(sq_acc[2 * t + 0]);
      float u_norm = sqrtf(sq_acc[2 * t + 1]);
      if (w_norm > 0.f && u_norm > 0.f) trust = w_norm / u_norm;
      if (trust_clip && trust > 1.f) trust = 1.f;
    }
    float pf = Elem<T>::to_f32(p[off]) - lr * trust * Elem<T>::to_f32(g[off]);
    p[off] = Elem<T>::from_f32(pf);
  }
}

template <typename scalar_t> struct ToHip2 { using type = float; };
template <> struct ToHip2<at::BFloat16> { using type = __hip_bfloat16; };
template <> struct ToHip2<at::Half> { using type = __half; };

}  // namespace

void multi_tensor_adamw(
    std::vector<at::Tensor> params, std::vector<at::Tensor> grads,
    std::vector<at::Tensor> exp_avgs, std::vector<at::Tensor> exp_avg_sqs,
    double lr, double beta1, double beta2, double eps, double wd, double bc1, double bc2) {
  TORCH_CHECK(!params.empty());
  auto table = build_table({params, grads, exp_avgs, exp_avg_sqs});
  auto stream = at::hip::getCurrentHIPStream();
  long blocks = std::min((long)4096, (table.total + kBlock - 1) / kBlock);
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, params[0].scalar_type(), "mt_adamw", [&] {
    using T = typename ToHip2<scalar_t>::type;
    hipLaunchKernelGGL((adamw_kernel<T>), dim3(blocks), dim3(kBlock), 0, stream,
        table.ptrs.data_ptr<long>(), table.prefix.data_ptr<long>(), table.n, table.total,
        (float)lr, (float)beta1, (float)beta2, (float)eps, (float)wd, (float)bc1, (float)bc2);
  });
  HIP_CHECK_LAST();
}

void multi_tensor_lerp(std::vector<at::Tensor> dsts, std::vector<at::Tensor> srcs, double weight) {
  TORCH_CHECK(!dsts.empty());
  auto table = build_table({dsts, srcs});
  auto stream = at::hip::getCurrentHIPStream();
  long blocks = std::min((long)4096, (table.total + kBlock - 1) / kBlock);
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, dsts[0].scalar_type(), "mt_lerp", [&] {
    using T = typename ToHip2<scalar_t>::type;
    hipLaunchKernelGGL((lerp_kernel<T>), dim3(blocks), dim3(kBlock), 0, stream,
        table.ptrs.data_ptr<long>(), table.prefix.data_ptr<long>(), table.n, table.total,
        (float)weight);
  });
  HIP_CHECK_LAST();
}

at::Tensor multi_tensor_l2norm(std::vector<at::Tensor> tensors) {
  TORCH_CHECK(!tensors.empty());
  auto table = build_table({tensors});
  auto out = at::zeros({}, tensors[0].options().dtype(at::kFloat));
  auto stream = at::hip::getCurrentHIPStream();
  long blocks = std::min((long)2048, (table.total + kBlock - 1) / kBlock);
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, tensors[0].scalar_type(), "mt_l2", [&] {
    using T = typename ToHip2<scalar_t>::type;
    hipLaunchKernelGGL((l2norm_kernel<T>), dim3(blocks), dim3(kBlock), 0, stream,
        table.ptrs.data_ptr<long>(), table.prefix.data_ptr<long>(), table.n, table.total,
        out.data_ptr<float>());
  });
  HIP_CHECK_LAST();
  return out.sqrt();
}


void multi_tensor_lamb(
    std::vector<at::Tensor> params, std::vector<at::Tensor> grads,
    std::vector<at::Tensor> exp_avgs, std::vector<at::Tensor> exp_avg_sqs,
    double lr, double beta1, double beta2, double beta3, double eps, double wd,
    double bc1, double bc2, double clip_norm, bool adapt, bool trust_clip) {
  TORCH_CHECK(!params.empty());
  auto table = build_table({params, grads, exp_avgs, exp_avg_sqs});
  auto sq_acc = at::zeros({table.n, 2}, params[0].options().dtype(at::kFloat));
  auto stream = at::hip::getCurrentHIPStream();
  long blocks = std::min((long)4096, (table.total + kBlock - 1) / kBlock);
  float inv_clip = clip_norm > 0 ? (float)(1.0 / clip_norm) : 1.f;
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, params[0].scalar_type(), "mt_lamb", [&] {
    using T = typename ToHip2<scalar_t>::type;
    hipLaunchKernelGGL((lamb_phase1_kernel<T>), dim3(blocks), dim3(kBlock), 0, stream,
        table.ptrs.data_ptr<long>(), table.prefix.data_ptr<long>(), table.n, table.total,
        (float)beta1, (float)beta2, (float)beta3, (float)eps, (float)wd,
        (float)bc1, (float)bc2, inv_clip, sq_acc.data_ptr<float>());
    hipLaunchKernelGGL((lamb_phase2_kernel<T>), dim3(blocks), dim3(kBlock), 0, stream,
        table.ptrs.data_ptr<long>(), table.prefix.data_ptr<long>(), table.n, table.total,
        (float)lr, adapt ? 1 : 0, trust_clip ? 1 : 0, sq_acc.data_ptr<float>());
  });
  HIP_CHECK_LAST();
}
