// Fused elementwise epilogues for gfx950:
//  * bias_act: y = act(x + b)  (GEMM bias+activation epilogue; replaces the
//    reference's Linear-bias + GELU pair, timm/layers/mlp.py:40-44)
//  * residual_scale_add: out = x + y * gamma[c] * keep[b]  (residual +
//    LayerScale + DropPath, vision_transformer.py:212-213)
// HBM-bound: vectorized 16B loads per lane (guide Appendix B, elementwise).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace {

constexpr int kBlock = 256;

enum ActId { kGelu = 0, kGeluTanh = 1, kSilu = 2, kRelu = 3, kIdentity = 4, kQuickGelu = 5 };

__device__ __forceinline__ float act_fwd(float x, int act) {
  switch (act) {
    case kGelu: return 0.5f * x * (1.f + erff(x * 0.70710678118654752f));
    case kGeluTanh: {
      float x3 = x * x * x;
      return 0.5f * x * (1.f + tanhf(0.7978845608028654f * (x + 0.044715f * x3)));
    }
    case kSilu: return x / (1.f + __expf(-x));
    case kRelu: return fmaxf(x, 0.f);
    case kQuickGelu: return x / (1.f + __expf(-1.702f * x));
    default: return x;
  }
}

__device__ __forceinline__ float act_bwd(float x, int act) {
  switch (act) {
    case kGelu: {
      float cdf = 0.5f * (1.f + erff(x * 0.70710678118654752f));
      float pdf = 0.3989422804014327f * __expf(-0.5f * x * x);
      return cdf + x * pdf;
    }
    case kGeluTanh: {
      float x2 = x * x;
      float inner = 0.7978845608028654f * (x + 0.044715f * x * x2);
      float t = tanhf(inner);
      float dinner = 0.7978845608028654f * (1.f + 3.f * 0.044715f * x2);
      return 0.5f * (1.f + t) + 0.5f * x * (1.f - t * t) * dinner;
    }
    case kSilu: {
      float s = 1.f / (1.f + __expf(-x));
      return s * (1.f + x * (1.f - s));
    }
    case kRelu: return x > 0.f ? 1.f : 0.f;
    case kQuickGelu: {
      float s = 1.f / (1.f + __expf(-1.702f * x));
      return s * (1.f + 1.702f * x * (1.f - s));
    }
    default: return 1.f;
  }
}

// ---- bias_act ----

template <typename T>
__global__ void bias_act_fwd_kernel(
    const T* __restrict__ x, const T* __restrict__ b, T* __restrict__ y,
    long n, int cols, int act) {
  long i0 = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  long stride = (long)gridDim.x * blockDim.x * 8;
  for (long i = i0; i < n; i += stride) {
    if (i + 8 <= n && (cols % 8 == 0)) {
      short8 xv = *reinterpret_cast<const short8*>(reinterpret_cast<const short*>(x) + i);
      int c = (int)(i % cols);
      short8 bv = *reinterpret_cast<const short8*>(reinterpret_cast<const short*>(b) + c);
      short8 out;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = Elem<T>::to_f32(reinterpret_cast<const T*>(&xv)[j]) +
                  Elem<T>::to_f32(reinterpret_cast<const T*>(&bv)[j]);
        reinterpret_cast<T*>(&out)[j] = Elem<T>::from_f32(act_fwd(f, act));
      }
      *reinterpret_cast<short8*>(reinterpret_cast<short*>(y) + i) = out;
    } else {
      for (long k = i; k < min(i + 8, n); ++k) {
        float f = Elem<T>::to_f32(x[k]) + Elem<T>::to_f32(b[k % cols]);
        y[k] = Elem<T>::from_f32(act_fwd(f, act));
      }
    }
  }
}

// Slow path (cols not 16B-divisible): per-element LDS atomics.
template <typename T>
__global__ void bias_act_bwd_kernel(
    const T* __restrict__ dy, const T* __restrict__ x, const T* __restrict__ b,
    T* __restrict__ dx, float* __restrict__ db, long n, int cols, int act) {
  extern __shared__ float db_s[];
  for (int i = threadIdx.x; i < cols; i += blockDim.x) db_s[i] = 0.f;
  __syncthreads();
  long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = i0; i < n; i += stride) {
    int c = (int)(i % cols);
    float f = Elem<T>::to_f32(x[i]) + Elem<T>::to_f32(b[c]);
    float g = Elem<T>::to_f32(dy[i]) * act_bwd(f, act);
    dx[i] = Elem<T>::from_f32(g);
    atomicAdd(&db_s[c], g);
  }
  __syncthreads();
  for (int i = threadIdx.x; i < cols; i += blockDim.x) atomicAdd(&db[i], db_s[i]);
}

// Fast path: wave-per-row, 16B vector loads, per-lane register dB accumulation
// (fixed lane->column mapping across rows), one global atomic/col/wave.
template <typename T, int kChunks>
__global__ __launch_bounds__(256)
void bias_act_bwd_fast_kernel(
    const T* __restrict__ dy, const T* __restrict__ x, const T* __restrict__ b,
    T* __restrict__ dx, float* __restrict__ db, long rows, int cols, int act) {
  const int wave = threadIdx.x / WAVE_SIZE;
  const int lane = threadIdx.x % WAVE_SIZE;
  const int nvec = cols / 8;
  constexpr int kWaves = 256 / WAVE_SIZE;

  float db_acc[kChunks][8];
  float b_reg[kChunks][8];
#pragma unroll
  for (int c = 0; c < kChunks; ++c) {
    int i = lane + c * WAVE_SIZE;
    if (i < nvec) {
      short8 bv = reinterpret_cast<const short8*>(b)[i];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        b_reg[c][j] = Elem<T>::to_f32(reinterpret_cast<const T*>(&bv)[j]);
        db_acc[c][j] = 0.f;
      }
    }
  }

  for (long row = blockIdx.x * kWaves + wave; row < rows; row += (long)gridDim.x * kWaves) {
    const short8* dyv = reinterpret_cast<const short8*>(dy + row * cols);
    const short8* xv = reinterpret_cast<const short8*>(x + row * cols);
    short8* dxv = reinterpret_cast<short8*>(dx + row * cols);
#pragma unroll
    for (int c = 0; c < kChunks; ++c) {
      int i = lane + c * WAVE_SIZE;
      if (i < nvec) {
        short8 dv = dyv[i];
        short8 xv8 = xv[i];
        short8 out;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float f = Elem<T>::to_f32(reinterpret_cast<const T*>(&xv8)[j]) + b_reg[c][j];
          float g = Elem<T>::to_f32(reinterpret_cast<const T*>(&dv)[j]) * act_bwd(f, act);
          reinterpret_cast<T*>(&out)[j] = Elem<T>::from_f32(g);
          db_acc[c][j] += g;
        }
        dxv[i] = out;
      }
    }
  }

  // block-level LDS reduction then one global atomic per col per block
  extern __shared__ float db_red[];
  for (int i = threadIdx.x; i < cols; i += 256) db_red[i] = 0.f;
  __syncthreads();
#pragma unroll
  for (int c = 0; c < kChunks; ++c) {
    int i = lane + c * WAVE_SIZE;
    if (i < nvec) {
#pragma unroll
      for (int j = 0; j < 8; ++j) atomicAdd(&db_red[i * 8 + j], db_acc[c][j]);
    }
  }
  __syncthreads();
  for (int i = threadIdx.x; i < cols; i += 256) atomicAdd(&db[i], db_red[i]);
}

// ---- residual_scale_add ----
// out = x + y * gamma[c] * keep[batch]; rows_per_batch = n / (B * cols)

template <typename T, bool kGamma, bool kKeep>
__global__ void residual_fwd_kernel(
    const T* __restrict__ x, const T* __restrict__ y,
    const T* __restrict__ gamma, const float* __restrict__ keep,
    T* __restrict__ out, long n, int cols, long per_batch) {
  long i0 = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  long stride = (long)gridDim.x * blockDim.x * 8;
  for (long i = i0; i < n; i += stride) {
    if (i + 8 <= n && (cols % 8 == 0)) {
      short8 xv = *reinterpret_cast<const short8*>(reinterpret_cast<const short*>(x) + i);
      short8 yv = *reinterpret_cast<const short8*>(reinterpret_cast<const short*>(y) + i);
      int c = (int)(i % cols);
      float kp = kKeep ? keep[i / per_batch] : 1.f;
      short8 gv;
      if (kGamma) gv = *reinterpret_cast<const short8*>(reinterpret_cast<const short*>(gamma) + c);
      short8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float yf = Elem<T>::to_f32(reinterpret_cast<const T*>(&yv)[j]);
        if (kGamma) yf *= Elem<T>::to_f32(reinterpret_cast<const T*>(&gv)[j]);
        float r = Elem<T>::to_f32(reinterpret_cast<const T*>(&xv)[j]) + yf * kp;
        reinterpret_cast<T*>(&o)[j] = Elem<T>::from_f32(r);
      }
      *reinterpret_cast<short8*>(reinterpret_cast<short*>(out) + i) = o;
    } else {
      for (long k = i; k < min(i + 8, n); ++k) {
        float yf = Elem<T>::to_f32(y[k]);
        if (kGamma) yf *= Elem<T>::to_f32(gamma[k % cols]);
        float kp = kKeep ? keep[k / per_batch] : 1.f;
        out[k] = Elem<T>::from_f32(Elem<T>::to_f32(x[k]) + yf * kp);
      }
    }
  }
}

template <typename T, bool kGamma, bool kKeep>
__global__ void residual_bwd_kernel(
    const T* __restrict__ dout, const T* __restrict__ y,
    const T* __restrict__ gamma, const float* __restrict__ keep,
    T* __restrict__ dy, float* __restrict__ dgamma,
    long n, int cols, long per_batch) {
  extern __shared__ float dg_s[];
  if (kGamma) {
    for (int i = threadIdx.x; i < cols; i += blockDim.x) dg_s[i] = 0.f;
    __syncthreads();
  }
  long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = i0; i < n; i += stride) {
    int c = (int)(i % cols);
    float g = Elem<T>::to_f32(dout[i]);
    float kp = kKeep ? keep[i / per_batch] : 1.f;
    float gk = g * kp;
    if (kGamma) {
      atomicAdd(&dg_s[c], gk * Elem<T>::to_f32(y[i]));
      dy[i] = Elem<T>::from_f32(gk * Elem<T>::to_f32(gamma[c]));
    } else {
      dy[i] = Elem<T>::from_f32(gk);
    }
  }
  if (kGamma) {
    __syncthreads();
    for (int i = threadIdx.x; i < cols; i += blockDim.x) atomicAdd(&dgamma[i], dg_s[i]);
  }
}

template <typename scalar_t> struct ToHip { using type = float; };
template <> struct ToHip<at::BFloat16> { using type = __hip_bfloat16; };
template <> struct ToHip<at::Half> { using type = __half; };

}  // namespace

at::Tensor bias_act_fwd(at::Tensor x, at::Tensor b, long act_id) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous());
  TORCH_CHECK(x.scalar_type() != at::kFloat || true);
  auto y = at::empty_like(x);
  long n = x.numel();
  int cols = x.size(-1);
  auto stream = at::hip::getCurrentHIPStream();
  int blocks = std::min((long)2048, (n + kBlock * 8 - 1) / (kBlock * 8));
  AT_DISPATCH_REDUCED_FLOATING_TYPES(x.scalar_type(), "bias_act_fwd", [&] {
    using T = typename ToHip<scalar_t>::type;
    hipLaunchKernelGGL((bias_act_fwd_kernel<T>), dim3(blocks), dim3(kBlock), 0, stream,
        (const T*)x.data_ptr(), (const T*)b.data_ptr(), (T*)y.data_ptr(), n, cols, (int)act_id);
  });
  HIP_CHECK_LAST();
  return y;
}

std::vector<at::Tensor> bias_act_bwd(at::Tensor dy, at::Tensor x, at::Tensor b, long act_id) {
  auto dx = at::empty_like(x);
  int cols = x.size(-1);
  auto db = at::zeros({cols}, x.options().dtype(at::kFloat));
  long n = x.numel();
  auto stream = at::hip::getCurrentHIPStream();
  AT_DISPATCH_REDUCED_FLOATING_TYPES(x.scalar_type(), "bias_act_bwd", [&] {
    using T = typename ToHip<scalar_t>::type;
    long rows = n / cols;
    int chunks = (cols / 8 + 63) / 64;
    if (cols % 8 == 0 && chunks <= 8) {
      int blocks = (int)std::min((long)640, (rows + 3) / 4);
      auto go = [&](auto tag) {
        hipLaunchKernelGGL((bias_act_bwd_fast_kernel<T, decltype(tag)::value>),
            dim3(blocks), dim3(256), cols * sizeof(float), stream,
            (const T*)dy.data_ptr(), (const T*)x.data_ptr(), (const T*)b.data_ptr(),
            (T*)dx.data_ptr(), db.data_ptr<float>(), rows, cols, (int)act_id);
      };
      if (chunks <= 1) go(std::integral_constant<int, 1>{});
      else if (chunks <= 2) go(std::integral_constant<int, 2>{});
      else if (chunks <= 4) go(std::integral_constant<int, 4>{});
      else if (chunks <= 6) go(std::integral_constant<int, 6>{});
      else go(std::integral_constant<int, 8>{});
    } else {
      int blocks = std::min((long)1024, (n + kBlock - 1) / kBlock);
      hipLaunchKernelGGL((bias_act_bwd_kernel<T>), dim3(blocks), dim3(kBlock),
          cols * sizeof(float), stream,
          (const T*)dy.data_ptr(), (const T*)x.data_ptr(), (const T*)b.data_ptr(),
          (T*)dx.data_ptr(), db.data_ptr<float>(), n, cols, (int)act_id);
    }
  });
  HIP_CHECK_LAST();
  return {dx, db.to(b.scalar_type())};
}

at::Tensor residual_scale_add_fwd(at::Tensor x, at::Tensor y,
                                  c10::optional<at::Tensor> gamma,
                                  c10::optional<at::Tensor> keep) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && y.is_contiguous());
  auto out = at::empty_like(x);
  long n = x.numel();
  int cols = x.size(-1);
  long per_batch = n / x.size(0);
  auto stream = at::hip::getCurrentHIPStream();
  int blocks = std::min((long)2048, (n + kBlock * 8 - 1) / (kBlock * 8));
  bool has_g = gamma.has_value();
  bool has_k = keep.has_value();
  AT_DISPATCH_REDUCED_FLOATING_TYPES(x.scalar_type(), "residual_fwd", [&] {
    using T = typename ToHip<scalar_t>::type;
    const T* gp = has_g ? (const T*)gamma->data_ptr() : nullptr;
    const float* kp = has_k ? keep->data_ptr<float>() : nullptr;
    auto launch = [&](auto gtag, auto ktag) {
      hipLaunchKernelGGL((residual_fwd_kernel<T, decltype(gtag)::value, decltype(ktag)::value>),
          dim3(blocks), dim3(kBlock), 0, stream,
          (const T*)x.data_ptr(), (const T*)y.data_ptr(), gp, kp,
          (T*)out.data_ptr(), n, cols, per_batch);
    };
    if (has_g && has_k) launch(std::true_type{}, std::true_type{});
    else if (has_g) launch(std::true_type{}, std::false_type{});
    else if (has_k) launch(std::false_type{}, std::true_type{});
    else launch(std::false_type{}, std::false_type{});
  });
  HIP_CHECK_LAST();
  return out;
}

std::vector<at::Tensor> residual_scale_add_bwd(at::Tensor dout,
                                               c10::optional<at::Tensor> y,
                                               c10::optional<at::Tensor> gamma,
                                               c10::optional<at::Tensor> keep) {
  auto dy = at::empty_like(dout);
  long n = dout.numel();
  int cols = dout.size(-1);
  long per_batch = n / dout.size(0);
  bool has_g = gamma.has_value();
  bool has_k = keep.has_value();
  at::Tensor dgamma;
  if (has_g) dgamma = at::zeros({cols}, dout.options().dtype(at::kFloat));
  auto stream = at::hip::getCurrentHIPStream();
  int blocks = std::min((long)1024, (n + kBlock - 1) / kBlock);
  AT_DISPATCH_REDUCED_FLOATING_TYPES(dout.scalar_type(), "residual_bwd", [&] {
    using T = typename ToHip<scalar_t>::type;
    const T* gp = has_g ? (const T*)gamma->data_ptr() : nullptr;
    const float* kp = has_k ? keep->data_ptr<float>() : nullptr;
    const T* yp = has_g ? (const T*)y->data_ptr() : nullptr;
    float* dgp = has_g ? dgamma.data_ptr<float>() : nullptr;
    size_t smem = has_g ? cols * sizeof(float) : 0;
    auto launch = [&](auto gtag, auto ktag) {
      hipLaunchKernelGGL((residual_bwd_kernel<T, decltype(gtag)::value, decltype(ktag)::value>),
          dim3(blocks), dim3(kBlock), smem, stream,
          (const T*)dout.data_ptr(), yp, gp, kp, (T*)dy.data_ptr(), dgp, n, cols, per_batch);
    };
    if (has_g && has_k) launch(std::true_type{}, std::true_type{});
    else if (has_g) launch(std::true_type{}, std::false_type{});
    else if (has_k) launch(std::false_type{}, std::true_type{});
    else launch(std::false_type{}, std::false_type{});
  });
  HIP_CHECK_LAST();
  at::Tensor dgamma_out;
  if (has_g) dgamma_out = dgamma.to(gamma->scalar_type());
  return {dy, dgamma_out};
}
