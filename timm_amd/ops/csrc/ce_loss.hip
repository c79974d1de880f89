// Fused cross-entropy loss for gfx950 (reference behavior:
// timm/loss/cross_entropy.py — LabelSmoothingCrossEntropy / SoftTargetCrossEntropy).
//
// One workgroup (4 waves / 256 threads) per row of logits. Forward computes
// the row max, exp-sum and target dot-product in a single HBM pass and emits
// per-row loss plus the log-sum-exp needed by backward. Backward recomputes
// softmax from the saved LSE (no [B,C] softmax tensor ever hits HBM) and
// writes dlogits in one pass.
//
//   hard labels:  loss = lse - (1-eps)*z_y - (eps/C) * sum_j z_j
//                 dlog = softmax - ((1-eps)*onehot_y + eps/C)
//   soft target:  loss = lse * sum_t - dot(t, z)
//                 dlog = softmax * sum_t - t
#include "common.h"
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

namespace {

constexpr int kThreads = 256;
constexpr int kWaves = kThreads / WAVE_SIZE;

// block-wide reductions through LDS (single __syncthreads round)
struct BlockRed {
  float lds[kWaves];
};

template <typename red_fn>
__device__ __forceinline__ float block_reduce(float v, float* lds, red_fn op, float init) {
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wave = threadIdx.x >> 6;
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = op(v, __shfl_xor(v, off, 64));
  if (lane == 0) lds[wave] = v;
  __syncthreads();
  float r = init;
#pragma unroll
  for (int w = 0; w < kWaves; ++w) r = op(r, lds[w]);
  return r;
}

struct MaxOp { __device__ float operator()(float a, float b) const { return fmaxf(a, b); } };
struct SumOp { __device__ float operator()(float a, float b) const { return a + b; } };

// ---------------- forward ----------------

template <typename T, bool kSoft>
__global__ void ce_fwd_kernel(
    const T* __restrict__ logits,      // [B, C]
    const int64_t* __restrict__ labels, // [B] (hard) or nullptr
    const T* __restrict__ soft,        // [B, C] (soft) or nullptr
    float* __restrict__ loss,          // [B]
    float* __restrict__ lse_out,       // [B]
    int C,
    float smoothing) {
  __shared__ float lds[kWaves];
  const int row = blockIdx.x;
  const T* z = logits + (int64_t)row * C;

  float vmax = -INFINITY;
  for (int c = threadIdx.x; c < C; c += kThreads)
    vmax = fmaxf(vmax, Elem<T>::to_f32(z[c]));
  vmax = block_reduce(vmax, lds, MaxOp(), -INFINITY);
  __syncthreads();

  // single pass: exp-sum, plain sum (for smoothing), target dot
  float esum = 0.f, zsum = 0.f, tdot = 0.f, tsum = 0.f;
  const T* t = kSoft ? soft + (int64_t)row * C : nullptr;
  for (int c = threadIdx.x; c < C; c += kThreads) {
    float v = Elem<T>::to_f32(z[c]);
    esum += __expf(v - vmax);
    if (kSoft) {
      float tv = Elem<T>::to_f32(t[c]);
      tdot += tv * v;
      tsum += tv;
    } else if (smoothing > 0.f) {
      zsum += v;
    }
  }
  esum = block_reduce(esum, lds, SumOp(), 0.f);
  __syncthreads();
  float lse = vmax + __logf(esum);

  if (kSoft) {
    tdot = block_reduce(tdot, lds, SumOp(), 0.f);
    __syncthreads();
    tsum = block_reduce(tsum, lds, SumOp(), 0.f);
    if (threadIdx.x == 0) {
      loss[row] = lse * tsum - tdot;
      lse_out[row] = lse;
    }
  } else {
    if (smoothing > 0.f) {
      zsum = block_reduce(zsum, lds, SumOp(), 0.f);
    }
    if (threadIdx.x == 0) {
      float zy = Elem<T>::to_f32(z[labels[row]]);
      float l = lse - (1.f - smoothing) * zy;
      if (smoothing > 0.f) l -= smoothing / (float)C * zsum;
      loss[row] = l;
      lse_out[row] = lse;
    }
  }
}

// ---------------- backward ----------------

template <typename T, bool kSoft>
__global__ void ce_bwd_kernel(
    const T* __restrict__ logits,
    const int64_t* __restrict__ labels,
    const T* __restrict__ soft,
    const float* __restrict__ lse,
    const float* __restrict__ dloss,   // [B] upstream grad per row
    T* __restrict__ dlogits,           // [B, C]
    int C,
    float smoothing) {
  const int row = blockIdx.x;
  const T* z = logits + (int64_t)row * C;
  T* dz = dlogits + (int64_t)row * C;
  const float row_lse = lse[row];
  const float g = dloss[row];

  float tsum = 1.f;
  const T* t = nullptr;
  if (kSoft) {
    t = soft + (int64_t)row * C;
    // recompute sum_t (cheap; usually exactly 1)
    __shared__ float lds[kWaves];
    float s = 0.f;
    for (int c = threadIdx.x; c < C; c += kThreads) s += Elem<T>::to_f32(t[c]);
    tsum = block_reduce(s, lds, SumOp(), 0.f);
    __syncthreads();
  }
  const int64_t y = kSoft ? -1 : labels[row];
  const float eps_c = smoothing / (float)C;

  for (int c = threadIdx.x; c < C; c += kThreads) {
    float p = __expf(Elem<T>::to_f32(z[c]) - row_lse);  // softmax
    float grad;
    if (kSoft) {
      grad = p * tsum - Elem<T>::to_f32(t[c]);
    } else {
      grad = p - eps_c - ((c == (int)y) ? (1.f - smoothing) : 0.f);
    }
    dz[c] = Elem<T>::from_f32(grad * g);
  }
}

}  // namespace

// returns (loss[B] fp32, lse[B] fp32)
std::vector<torch::Tensor> ce_loss_fwd(
    torch::Tensor logits,
    torch::Tensor target,   // int64 [B] or float/bf16 [B, C]
    double smoothing) {
  TORCH_CHECK(logits.is_cuda() && logits.dim() == 2, "logits must be CUDA [B, C]");
  logits = logits.contiguous();
  const int B = logits.size(0), C = logits.size(1);
  auto opts = logits.options().dtype(torch::kFloat32);
  auto loss = torch::empty({B}, opts);
  auto lse = torch::empty({B}, opts);
  const bool soft = target.dim() == 2;
  if (soft) {
    TORCH_CHECK(target.scalar_type() == logits.scalar_type(), "soft target dtype must match logits");
    target = target.contiguous();
  } else {
    TORCH_CHECK(target.scalar_type() == torch::kInt64, "hard labels must be int64");
    target = target.contiguous();
  }
  auto stream = at::hip::getCurrentHIPStream();
  dim3 grid(B), block(kThreads);

  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, logits.scalar_type(), "ce_fwd", [&] {
    using T = std::conditional_t<std::is_same_v<scalar_t, at::BFloat16>, __hip_bfloat16,
              std::conditional_t<std::is_same_v<scalar_t, at::Half>, __half, float>>;
    if (soft) {
      hipLaunchKernelGGL((ce_fwd_kernel<T, true>), grid, block, 0, stream,
          (const T*)logits.data_ptr(), nullptr, (const T*)target.data_ptr(),
          loss.data_ptr<float>(), lse.data_ptr<float>(), C, (float)smoothing);
    } else {
      hipLaunchKernelGGL((ce_fwd_kernel<T, false>), grid, block, 0, stream,
          (const T*)logits.data_ptr(), target.data_ptr<int64_t>(), nullptr,
          loss.data_ptr<float>(), lse.data_ptr<float>(), C, (float)smoothing);
    }
  });
  HIP_CHECK_LAST();
  return {loss, lse};
}

torch::Tensor ce_loss_bwd(
    torch::Tensor logits,
    torch::Tensor target,
    torch::Tensor lse,
    torch::Tensor dloss,    // [B] fp32
    double smoothing) {
  logits = logits.contiguous();
  const int B = logits.size(0), C = logits.size(1);
  auto dlogits = torch::empty_like(logits);
  const bool soft = target.dim() == 2;
  target = target.contiguous();
  dloss = dloss.contiguous().to(torch::kFloat32);
  auto stream = at::hip::getCurrentHIPStream();
  dim3 grid(B), block(kThreads);

  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, logits.scalar_type(), "ce_bwd", [&] {
    using T = std::conditional_t<std::is_same_v<scalar_t, at::BFloat16>, __hip_bfloat16,
              std::conditional_t<std::is_same_v<scalar_t, at::Half>, __half, float>>;
    if (soft) {
      hipLaunchKernelGGL((ce_bwd_kernel<T, true>), grid, block, 0, stream,
          (const T*)logits.data_ptr(), nullptr, (const T*)target.data_ptr(),
          lse.data_ptr<float>(), dloss.data_ptr<float>(), (T*)dlogits.data_ptr(),
          C, (float)smoothing);
    } else {
      hipLaunchKernelGGL((ce_bwd_kernel<T, false>), grid, block, 0, stream,
          (const T*)logits.data_ptr(), target.data_ptr<int64_t>(), nullptr,
          lse.data_ptr<float>(), dloss.data_ptr<float>(), (T*)dlogits.data_ptr(),
          C, (float)smoothing);
    }
  });
  HIP_CHECK_LAST();
  return dlogits;
}
