// Data-path kernels for gfx950: fused uint8 -> normalized-tensor conversion
// (loader prefetch; replaces the reference's .float().sub_(mean).div_(std)
// 3-kernel chain, timm/data/loader.py:116) and masked global pooling for
// NaFlex batches (timm/models/naflexvit.py:1065).
#include "common.h"
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

namespace {

// ---------------- fused uint8 normalize ----------------
// x: [B, C, H, W] uint8 contiguous. out: same shape, (x - mean[c]) * inv_std[c].
// Bandwidth-bound: each thread converts 4 pixels (uchar4 load, 4-wide store).

template <typename T>
__global__ void u8_normalize_kernel(
    const unsigned char* __restrict__ x,
    T* __restrict__ out,
    const float* __restrict__ mean,     // [C]
    const float* __restrict__ inv_std,  // [C]
    int64_t plane,                      // H*W
    int C,
    int64_t total4) {                   // ceil(B*C*H*W / 4)
  int64_t i4 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i4 >= total4) return;
  int64_t i = i4 * 4;

  // all 4 pixels share a channel unless the boundary falls inside the quad;
  // plane (H*W) is almost always a multiple of 4, so take the fast path
  uchar4 v = *reinterpret_cast<const uchar4*>(x + i);
  int c0 = (int)((i / plane) % C);
  if (((i & 3) == 0) && ((plane & 3) == 0)) {
    float m = mean[c0], s = inv_std[c0];
    out[i + 0] = Elem<T>::from_f32(((float)v.x - m) * s);
    out[i + 1] = Elem<T>::from_f32(((float)v.y - m) * s);
    out[i + 2] = Elem<T>::from_f32(((float)v.z - m) * s);
    out[i + 3] = Elem<T>::from_f32(((float)v.w - m) * s);
  } else {
    const unsigned char px[4] = {v.x, v.y, v.z, v.w};
#pragma unroll
    for (int k = 0; k < 4; ++k) {
      int64_t j = i + k;
      int c = (int)((j / plane) % C);
      out[j] = Elem<T>::from_f32(((float)px[k] - mean[c]) * inv_std[c]);
    }
  }
}

// ---------------- masked global pooling ----------------
// x: [B, N, C] (prefix tokens already stripped by the host), valid: [B, N]
// (uint8/bool). One workgroup per (row-block of C, batch); threads stride C.
// avg:  out[b, c] = sum_n valid * x / max(1, count)
// max:  out[b, c] = max over valid (argmax saved for backward)

constexpr int kPoolThreads = 256;

template <typename T, bool kMax>
__global__ void masked_pool_fwd_kernel(
    const T* __restrict__ x,
    const unsigned char* __restrict__ valid,
    T* __restrict__ out,         // [B, C]
    int* __restrict__ argmax,    // [B, C] (max only)
    float* __restrict__ count,   // [B]
    int N, int C) {
  const int b = blockIdx.y;
  const int c = blockIdx.x * kPoolThreads + threadIdx.x;
  if (c >= C) return;
  const T* xb = x + (int64_t)b * N * C;
  const unsigned char* vb = valid + (int64_t)b * N;

  if (kMax) {
    float best = -INFINITY;
    int best_n = 0;
    for (int n = 0; n < N; ++n) {
      if (!vb[n]) continue;
      float v = Elem<T>::to_f32(xb[(int64_t)n * C + c]);
      if (v > best) { best = v; best_n = n; }
    }
    out[(int64_t)b * C + c] = Elem<T>::from_f32(best);
    argmax[(int64_t)b * C + c] = best_n;
  } else {
    float s = 0.f;
    int cnt = 0;
    for (int n = 0; n < N; ++n) {
      if (!vb[n]) continue;
      s += Elem<T>::to_f32(xb[(int64_t)n * C + c]);
      ++cnt;
    }
    float denom = (float)max(cnt, 1);
    out[(int64_t)b * C + c] = Elem<T>::from_f32(s / denom);
    if (c == 0) count[b] = denom;
  }
}

template <typename T, bool kMax>
__global__ void masked_pool_bwd_kernel(
    const T* __restrict__ dy,    // [B, C]
    const unsigned char* __restrict__ valid,
    const int* __restrict__ argmax,
    const float* __restrict__ count,
    T* __restrict__ dx,          // [B, N, C] (pre-zeroed for max)
    int N, int C) {
  const int b = blockIdx.y;
  const int c = blockIdx.x * kPoolThreads + threadIdx.x;
  if (c >= C) return;
  const float g = Elem<T>::to_f32(dy[(int64_t)b * C + c]);
  T* dxb = dx + (int64_t)b * N * C;
  if (kMax) {
    dxb[(int64_t)argmax[(int64_t)b * C + c] * C + c] = Elem<T>::from_f32(g);
  } else {
    const unsigned char* vb = valid + (int64_t)b * N;
    const float gs = g / count[b];
    for (int n = 0; n < N; ++n)
      dxb[(int64_t)n * C + c] = vb[n] ? Elem<T>::from_f32(gs) : Elem<T>::from_f32(0.f);
  }
}

}  // namespace

torch::Tensor u8_normalize(
    torch::Tensor x,          // [B, C, H, W] uint8
    torch::Tensor mean,       // [C] float
    torch::Tensor inv_std,    // [C] float
    at::ScalarType out_dtype) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kUInt8, "x must be CUDA uint8");
  TORCH_CHECK(x.dim() == 4, "x must be [B, C, H, W]");
  x = x.contiguous();
  mean = mean.contiguous().to(torch::kFloat32);
  inv_std = inv_std.contiguous().to(torch::kFloat32);
  const int C = x.size(1);
  const int64_t plane = (int64_t)x.size(2) * x.size(3);
  const int64_t total = x.numel();
  TORCH_CHECK(total % 4 == 0, "numel must be a multiple of 4");
  auto out = torch::empty_like(x, x.options().dtype(out_dtype));
  const int64_t total4 = total / 4;
  const int threads = 256;
  const int64_t blocks = (total4 + threads - 1) / threads;
  auto stream = at::hip::getCurrentHIPStream();

  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, out_dtype, "u8_norm", [&] {
    using T = std::conditional_t<std::is_same_v<scalar_t, at::BFloat16>, __hip_bfloat16,
              std::conditional_t<std::is_same_v<scalar_t, at::Half>, __half, float>>;
    hipLaunchKernelGGL((u8_normalize_kernel<T>), dim3(blocks), dim3(threads), 0, stream,
        x.data_ptr<unsigned char>(), (T*)out.data_ptr(),
        mean.data_ptr<float>(), inv_std.data_ptr<float>(), plane, C, total4);
  });
  HIP_CHECK_LAST();
  return out;
}

// returns (out[B,C], argmax[B,C] int32 or count[B] f32)
std::vector<torch::Tensor> masked_pool_fwd(
    torch::Tensor x,        // [B, N, C]
    torch::Tensor valid,    // [B, N] bool/uint8
    bool is_max) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 3, "x must be CUDA [B, N, C]");
  x = x.contiguous();
  valid = valid.contiguous().to(torch::kUInt8);
  const int B = x.size(0), N = x.size(1), C = x.size(2);
  auto out = torch::empty({B, C}, x.options());
  auto argmax = is_max ? torch::empty({B, C}, x.options().dtype(torch::kInt32))
                       : torch::empty({0}, x.options().dtype(torch::kInt32));
  auto count = is_max ? torch::empty({0}, x.options().dtype(torch::kFloat32))
                      : torch::empty({B}, x.options().dtype(torch::kFloat32));
  dim3 grid(cdiv(C, kPoolThreads), B), block(kPoolThreads);
  auto stream = at::hip::getCurrentHIPStream();

  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, x.scalar_type(), "mpool_fwd", [&] {
    using T = std::conditional_t<std::is_same_v<scalar_t, at::BFloat16>, __hip_bfloat16,
              std::conditional_t<std::is_same_v<scalar_t, at::Half>, __half, float>>;
    if (is_max) {
      hipLaunchKernelGGL((masked_pool_fwd_kernel<T, true>), grid, block, 0, stream,
          (const T*)x.data_ptr(), valid.data_ptr<unsigned char>(), (T*)out.data_ptr(),
          argmax.data_ptr<int>(), nullptr, N, C);
    } else {
      hipLaunchKernelGGL((masked_pool_fwd_kernel<T, false>), grid, block, 0, stream,
          (const T*)x.data_ptr(), valid.data_ptr<unsigned char>(), (T*)out.data_ptr(),
          nullptr, count.data_ptr<float>(), N, C);
    }
  });
  HIP_CHECK_LAST();
  return {out, is_max ? argmax : count};
}

torch::Tensor masked_pool_bwd(
    torch::Tensor dy,       // [B, C]
    torch::Tensor valid,    // [B, N]
    torch::Tensor aux,      // argmax (max) or count (avg)
    long n_tokens,
    bool is_max) {
  dy = dy.contiguous();
  valid = valid.contiguous().to(torch::kUInt8);
  const int B = dy.size(0), C = dy.size(1), N = (int)n_tokens;
  auto dx = is_max ? torch::zeros({B, N, C}, dy.options())
                   : torch::empty({B, N, C}, dy.options());
  dim3 grid(cdiv(C, kPoolThreads), B), block(kPoolThreads);
  auto stream = at::hip::getCurrentHIPStream();

  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, dy.scalar_type(), "mpool_bwd", [&] {
    using T = std::conditional_t<std::is_same_v<scalar_t, at::BFloat16>, __hip_bfloat16,
              std::conditional_t<std::is_same_v<scalar_t, at::Half>, __half, float>>;
    if (is_max) {
      hipLaunchKernelGGL((masked_pool_bwd_kernel<T, true>), grid, block, 0, stream,
          (const T*)dy.data_ptr(), valid.data_ptr<unsigned char>(),
          aux.data_ptr<int>(), nullptr, (T*)dx.data_ptr(), N, C);
    } else {
      hipLaunchKernelGGL((masked_pool_bwd_kernel<T, false>), grid, block, 0, stream,
          (const T*)dy.data_ptr(), valid.data_ptr<unsigned char>(),
          nullptr, aux.data_ptr<float>(), (T*)dx.data_ptr(), N, C);
    }
  });
  HIP_CHECK_LAST();
  return dx;
}
