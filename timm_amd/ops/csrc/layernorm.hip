// Fused LayerNorm / RMSNorm forward + backward for gfx950.
//
// Replaces the reference's F.layer_norm / F.rms_norm hot path
// (timm/layers/norm.py:70-290, fast_norm.py:119-160).
// One wave per row (4 rows per 256-thread block), fp32 accumulation,
// vectorized 16B (8 x bf16) loads per lane (guide §Guideline 13: scalar bf16
// loads are ~2x slower). dW/dB reduced via fp32 atomics.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace {

constexpr int kBlock = 256;
constexpr int kWavesPerBlock = kBlock / WAVE_SIZE;

// ----------------------------------------------------------------------------
// forward: y = (x - mean) * rstd * w + b ; saves mean, rstd (fp32 per row)
// RMS variant: y = x * rstd * w (mean==0 fixed), saves rstd only.
// ----------------------------------------------------------------------------

template <typename T, bool kRms>
__global__ void norm_fwd_kernel(
    const T* __restrict__ x,
    const T* __restrict__ w,
    const T* __restrict__ b,
    T* __restrict__ y,
    float* __restrict__ mean_out,
    float* __restrict__ rstd_out,
    int rows, int cols, float eps) {
  const int wave = threadIdx.x / WAVE_SIZE;
  const int lane = threadIdx.x % WAVE_SIZE;
  const bool vec_ok = (cols % 8 == 0);

  for (int row = blockIdx.x * kWavesPerBlock + wave; row < rows;
       row += gridDim.x * kWavesPerBlock) {
    const T* xr = x + (long)row * cols;
    T* yr = y + (long)row * cols;

    float sum = 0.f, sumsq = 0.f;
    if (vec_ok && sizeof(T) == 2) {
      const short8* xv = reinterpret_cast<const short8*>(xr);
      int nvec = cols / 8;
      for (int i = lane; i < nvec; i += WAVE_SIZE) {
        short8 v = xv[i];
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float f = Elem<T>::to_f32(reinterpret_cast<const T*>(&v)[j]);
          sum += f;
          sumsq += f * f;
        }
      }
    } else {
      for (int i = lane; i < cols; i += WAVE_SIZE) {
        float f = Elem<T>::to_f32(xr[i]);
        sum += f;
        sumsq += f * f;
      }
    }
    sum = wave_reduce_sum(sum);
    sumsq = wave_reduce_sum(sumsq);

    float mean = kRms ? 0.f : sum / cols;
    float var = sumsq / cols - mean * mean;
    float rstd = rsqrtf(var + eps);
    if (lane == 0) {
      if (!kRms && mean_out) mean_out[row] = mean;
      rstd_out[row] = rstd;
    }

    if (vec_ok && sizeof(T) == 2) {
      const short8* xv = reinterpret_cast<const short8*>(xr);
      const short8* wv = reinterpret_cast<const short8*>(w);
      const short8* bv = b ? reinterpret_cast<const short8*>(b) : nullptr;
      short8* yv = reinterpret_cast<short8*>(yr);
      int nvec = cols / 8;
      for (int i = lane; i < nvec; i += WAVE_SIZE) {
        short8 v = xv[i];
        short8 wg = wv[i];
        short8 bb;
        if (bv) bb = bv[i];
        short8 out;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float f = Elem<T>::to_f32(reinterpret_cast<const T*>(&v)[j]);
          float wf = Elem<T>::to_f32(reinterpret_cast<const T*>(&wg)[j]);
          float o = (f - mean) * rstd * wf;
          if (bv) o += Elem<T>::to_f32(reinterpret_cast<const T*>(&bb)[j]);
          reinterpret_cast<T*>(&out)[j] = Elem<T>::from_f32(o);
        }
        yv[i] = out;
      }
    } else {
      for (int i = lane; i < cols; i += WAVE_SIZE) {
        float f = Elem<T>::to_f32(xr[i]);
        float o = (f - mean) * rstd * Elem<T>::to_f32(w[i]);
        if (b) o += Elem<T>::to_f32(b[i]);
        yr[i] = Elem<T>::from_f32(o);
      }
    }
  }
}

// ----------------------------------------------------------------------------
// backward.
// xhat = (x - mean) * rstd
// dx = rstd * (dy*w - mean_c(dy*w) - xhat * mean_c(dy*w*xhat))   [LN]
// dx = rstd * (dy*w - xhat * mean_c(dy*w*xhat))                  [RMS]
// dw += dy * xhat ; db += dy   (fp32 atomics into scratch buffers)
// ----------------------------------------------------------------------------

template <typename T, bool kRms>
__global__ void norm_bwd_kernel(
    const T* __restrict__ dy,
    const T* __restrict__ x,
    const T* __restrict__ w,
    const float* __restrict__ mean_in,
    const float* __restrict__ rstd_in,
    T* __restrict__ dx,
    float* __restrict__ dw,  // [cols] fp32, pre-zeroed
    float* __restrict__ db,  // [cols] fp32, pre-zeroed (LN only)
    int rows, int cols) {
  const int wave = threadIdx.x / WAVE_SIZE;
  const int lane = threadIdx.x % WAVE_SIZE;

  // per-block fp32 LDS accumulators for dw/db reduce across this block's rows
  extern __shared__ float smem[];
  float* dw_s = smem;              // [cols]
  float* db_s = kRms ? nullptr : smem + cols;  // [cols]
  for (int i = threadIdx.x; i < cols; i += blockDim.x) {
    dw_s[i] = 0.f;
    if (!kRms) db_s[i] = 0.f;
  }
  __syncthreads();

  for (int row = blockIdx.x * kWavesPerBlock + wave; row < rows;
       row += gridDim.x * kWavesPerBlock) {
    const T* dyr = dy + (long)row * cols;
    const T* xr = x + (long)row * cols;
    T* dxr = dx + (long)row * cols;
    const float mean = kRms ? 0.f : mean_in[row];
    const float rstd = rstd_in[row];

    // pass 1: c1 = sum(dy*w*xhat), c2 = sum(dy*w)
    float c1 = 0.f, c2 = 0.f;
    for (int i = lane; i < cols; i += WAVE_SIZE) {
      float g = Elem<T>::to_f32(dyr[i]) * Elem<T>::to_f32(w[i]);
      float xhat = (Elem<T>::to_f32(xr[i]) - mean) * rstd;
      c1 += g * xhat;
      c2 += g;
    }
    c1 = wave_reduce_sum(c1) / cols;
    c2 = wave_reduce_sum(c2) / cols;

    // pass 2: dx + accumulate dw/db into LDS
    for (int i = lane; i < cols; i += WAVE_SIZE) {
      float gdy = Elem<T>::to_f32(dyr[i]);
      float g = gdy * Elem<T>::to_f32(w[i]);
      float xhat = (Elem<T>::to_f32(xr[i]) - mean) * rstd;
      float v = kRms ? (g - xhat * c1) : (g - c2 - xhat * c1);
      dxr[i] = Elem<T>::from_f32(rstd * v);
      atomicAdd(&dw_s[i], gdy * xhat);
      if (!kRms) atomicAdd(&db_s[i], gdy);
    }
  }

  __syncthreads();
  for (int i = threadIdx.x; i < cols; i += blockDim.x) {
    atomicAdd(&dw[i], dw_s[i]);
    if (!kRms) atomicAdd(&db[i], db_s[i]);
  }
}

// Fast backward: vectorized 16B loads, per-lane register accumulation of
// dW/dB (lane->chunk mapping is fixed across rows), one global fp32 atomic
// per column per wave at the end.  Requires cols % 8 == 0 and
// cols <= kChunks*512.
template <typename T, bool kRms, int kChunks>
__global__ __launch_bounds__(kBlock)
void norm_bwd_fast_kernel(
    const T* __restrict__ dy,
    const T* __restrict__ x,
    const T* __restrict__ w,
    const float* __restrict__ mean_in,
    const float* __restrict__ rstd_in,
    T* __restrict__ dx,
    float* __restrict__ dw,
    float* __restrict__ db,
    int rows, int cols) {
  const int wave = threadIdx.x / WAVE_SIZE;
  const int lane = threadIdx.x % WAVE_SIZE;
  const int nvec = cols / 8;

  float dw_acc[kChunks][8];
  float db_acc[kChunks][8];  // folded away for kRms (all uses guarded by constexpr cond)
#pragma unroll
  for (int c = 0; c < kChunks; ++c)
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      dw_acc[c][j] = 0.f;
      if (!kRms) db_acc[c][j] = 0.f;
    }

  // preload w into registers (shared across all rows)
  float w_reg[kChunks][8];
#pragma unroll
  for (int c = 0; c < kChunks; ++c) {
    int i = lane + c * WAVE_SIZE;
    if (i < nvec) {
      short8 wv = reinterpret_cast<const short8*>(w)[i];
#pragma unroll
      for (int j = 0; j < 8; ++j) w_reg[c][j] = Elem<T>::to_f32(reinterpret_cast<const T*>(&wv)[j]);
    }
  }

  for (int row = blockIdx.x * kWavesPerBlock + wave; row < rows;
       row += gridDim.x * kWavesPerBlock) {
    const short8* dyv = reinterpret_cast<const short8*>(dy + (long)row * cols);
    const short8* xv = reinterpret_cast<const short8*>(x + (long)row * cols);
    const float mean = kRms ? 0.f : mean_in[row];
    const float rstd = rstd_in[row];

    float dy_reg[kChunks][8], xh_reg[kChunks][8];
    float c1 = 0.f, c2 = 0.f;
#pragma unroll
    for (int c = 0; c < kChunks; ++c) {
      int i = lane + c * WAVE_SIZE;
      if (i < nvec) {
        short8 dv = dyv[i];
        short8 xv8 = xv[i];
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float g = Elem<T>::to_f32(reinterpret_cast<const T*>(&dv)[j]);
          float xhat = (Elem<T>::to_f32(reinterpret_cast<const T*>(&xv8)[j]) - mean) * rstd;
          dy_reg[c][j] = g;
          xh_reg[c][j] = xhat;
          float gw = g * w_reg[c][j];
          c1 += gw * xhat;
          c2 += gw;
        }
      }
    }
    c1 = wave_reduce_sum(c1) / cols;
    c2 = wave_reduce_sum(c2) / cols;

    short8* dxv = reinterpret_cast<short8*>(dx + (long)row * cols);
#pragma unroll
    for (int c = 0; c < kChunks; ++c) {
      int i = lane + c * WAVE_SIZE;
      if (i < nvec) {
        short8 out;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float g = dy_reg[c][j];
          float xhat = xh_reg[c][j];
          float gw = g * w_reg[c][j];
          float val = kRms ? (gw - xhat * c1) : (gw - c2 - xhat * c1);
          reinterpret_cast<T*>(&out)[j] = Elem<T>::from_f32(rstd * val);
          dw_acc[c][j] += g * xhat;
          if (!kRms) db_acc[c][j] += g;
        }
        dxv[i] = out;
      }
    }
  }

  // flush: block-level LDS reduction (4-way wave contention only), then ONE
  // global atomic per column per block — global atomic chains are gridDim
  // deep instead of n_waves deep (same-address fp32 atomics serialize in L2).
  extern __shared__ float red_s[];  // [cols] dw (+ [cols] db for LN)
  float* dw_s = red_s;
  float* db_s = red_s + cols;
  for (int i = threadIdx.x; i < cols; i += kBlock) {
    dw_s[i] = 0.f;
    if (!kRms) db_s[i] = 0.f;
  }
  __syncthreads();
#pragma unroll
  for (int c = 0; c < kChunks; ++c) {
    int i = lane + c * WAVE_SIZE;
    if (i < nvec) {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        atomicAdd(&dw_s[i * 8 + j], dw_acc[c][j]);
        if (!kRms) atomicAdd(&db_s[i * 8 + j], db_acc[c][j]);
      }
    }
  }
  __syncthreads();
  for (int i = threadIdx.x; i < cols; i += kBlock) {
    atomicAdd(&dw[i], dw_s[i]);
    if (!kRms) atomicAdd(&db[i], db_s[i]);
  }
}

template <typename T>
void launch_norm_fwd(const at::Tensor& x, const at::Tensor& w, const c10::optional<at::Tensor>& b,
                     at::Tensor& y, at::Tensor& mean, at::Tensor& rstd, double eps, bool rms,
                     int rows, int cols, hipStream_t stream) {
  int blocks = std::min(cdiv(rows, kWavesPerBlock), 8192);
  if (rms) {
    hipLaunchKernelGGL((norm_fwd_kernel<T, true>), dim3(blocks), dim3(kBlock), 0, stream,
        (const T*)x.data_ptr(), (const T*)w.data_ptr(),
        b.has_value() ? (const T*)b->data_ptr() : nullptr,
        (T*)y.data_ptr(), nullptr, rstd.data_ptr<float>(), rows, cols, (float)eps);
  } else {
    hipLaunchKernelGGL((norm_fwd_kernel<T, false>), dim3(blocks), dim3(kBlock), 0, stream,
        (const T*)x.data_ptr(), (const T*)w.data_ptr(),
        b.has_value() ? (const T*)b->data_ptr() : nullptr,
        (T*)y.data_ptr(), mean.data_ptr<float>(), rstd.data_ptr<float>(), rows, cols, (float)eps);
  }
  HIP_CHECK_LAST();
}

template <typename T, bool kRms>
bool try_launch_norm_bwd_fast(const T* dy, const T* x, const T* w,
                              const float* mean, const float* rstd,
                              T* dx, float* dw, float* db,
                              int rows, int cols, hipStream_t stream) {
  if (cols % 8 != 0 || sizeof(T) != 2) return false;
  // 320 blocks ≈ 1.25 waves/CU keeps the chip fed for a bandwidth-bound
  // kernel while bounding the global-atomic chain depth for dW/dB
  int blocks = std::min(cdiv(rows, kWavesPerBlock), 320);
  int chunks = cdiv(cols / 8, WAVE_SIZE);
  size_t smem = (kRms ? 1 : 2) * cols * sizeof(float);
  auto go = [&](auto tag) {
    constexpr int kChunks = decltype(tag)::value;
    hipLaunchKernelGGL((norm_bwd_fast_kernel<T, kRms, kChunks>),
        dim3(blocks), dim3(kBlock), smem, stream,
        dy, x, w, mean, rstd, dx, dw, db, rows, cols);
  };
  if (chunks <= 1) go(std::integral_constant<int, 1>{});
  else if (chunks <= 2) go(std::integral_constant<int, 2>{});
  else if (chunks <= 4) go(std::integral_constant<int, 4>{});
  else if (chunks <= 8 && kRms) go(std::integral_constant<int, 8>{});
  else if (chunks <= 6) go(std::integral_constant<int, 6>{});
  else return false;
  HIP_CHECK_LAST();
  return true;
}

template <typename T>
void launch_norm_bwd(const at::Tensor& dy, const at::Tensor& x, const at::Tensor& w,
                     const c10::optional<at::Tensor>& mean, const at::Tensor& rstd,
                     at::Tensor& dx, at::Tensor& dw, at::Tensor& db, bool rms,
                     int rows, int cols, hipStream_t stream) {
  if (rms) {
    if (try_launch_norm_bwd_fast<T, true>(
            (const T*)dy.data_ptr(), (const T*)x.data_ptr(), (const T*)w.data_ptr(),
            nullptr, rstd.data_ptr<float>(), (T*)dx.data_ptr(),
            dw.data_ptr<float>(), nullptr, rows, cols, stream))
      return;
  } else {
    if (try_launch_norm_bwd_fast<T, false>(
            (const T*)dy.data_ptr(), (const T*)x.data_ptr(), (const T*)w.data_ptr(),
            mean->data_ptr<float>(), rstd.data_ptr<float>(), (T*)dx.data_ptr(),
            dw.data_ptr<float>(), db.data_ptr<float>(), rows, cols, stream))
      return;
  }
  int blocks = std::min(cdiv(rows, kWavesPerBlock), 1024);
  size_t smem = (rms ? 1 : 2) * cols * sizeof(float);
  if (rms) {
    hipLaunchKernelGGL((norm_bwd_kernel<T, true>), dim3(blocks), dim3(kBlock), smem, stream,
        (const T*)dy.data_ptr(), (const T*)x.data_ptr(), (const T*)w.data_ptr(),
        nullptr, rstd.data_ptr<float>(),
        (T*)dx.data_ptr(), dw.data_ptr<float>(), nullptr, rows, cols);
  } else {
    hipLaunchKernelGGL((norm_bwd_kernel<T, false>), dim3(blocks), dim3(kBlock), smem, stream,
        (const T*)dy.data_ptr(), (const T*)x.data_ptr(), (const T*)w.data_ptr(),
        mean->data_ptr<float>(), rstd.data_ptr<float>(),
        (T*)dx.data_ptr(), dw.data_ptr<float>(), db.data_ptr<float>(), rows, cols);
  }
  HIP_CHECK_LAST();
}

}  // namespace

std::vector<at::Tensor> layer_norm_fwd(at::Tensor x, at::Tensor w, at::Tensor b, double eps) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous());
  int cols = x.size(-1);
  long rows = x.numel() / cols;
  auto y = at::empty_like(x);
  auto mean = at::empty({rows}, x.options().dtype(at::kFloat));
  auto rstd = at::empty({rows}, x.options().dtype(at::kFloat));
  auto stream = at::hip::getCurrentHIPStream();
  c10::optional<at::Tensor> bias(b);
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, x.scalar_type(), "layer_norm_fwd", [&] {
    using T = std::conditional_t<std::is_same_v<scalar_t, at::BFloat16>, __hip_bfloat16,
              std::conditional_t<std::is_same_v<scalar_t, at::Half>, __half, float>>;
    launch_norm_fwd<T>(x, w, bias, y, mean, rstd, eps, false, rows, cols, stream);
  });
  return {y, mean, rstd};
}

std::vector<at::Tensor> layer_norm_bwd(at::Tensor dy, at::Tensor x, at::Tensor w,
                                       at::Tensor mean, at::Tensor rstd) {
  int cols = x.size(-1);
  long rows = x.numel() / cols;
  auto dx = at::empty_like(x);
  auto dw = at::zeros({cols}, x.options().dtype(at::kFloat));
  auto db = at::zeros({cols}, x.options().dtype(at::kFloat));
  auto stream = at::hip::getCurrentHIPStream();
  c10::optional<at::Tensor> mean_opt(mean);
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, x.scalar_type(), "layer_norm_bwd", [&] {
    using T = std::conditional_t<std::is_same_v<scalar_t, at::BFloat16>, __hip_bfloat16,
              std::conditional_t<std::is_same_v<scalar_t, at::Half>, __half, float>>;
    launch_norm_bwd<T>(dy, x, w, mean_opt, rstd, dx, dw, db, false, rows, cols, stream);
  });
  return {dx, dw.to(w.scalar_type()), db.to(w.scalar_type())};
}

std::vector<at::Tensor> rms_norm_fwd(at::Tensor x, at::Tensor w, double eps) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous());
  int cols = x.size(-1);
  long rows = x.numel() / cols;
  auto y = at::empty_like(x);
  auto rstd = at::empty({rows}, x.options().dtype(at::kFloat));
  auto mean = at::Tensor();
  auto stream = at::hip::getCurrentHIPStream();
  c10::optional<at::Tensor> bias;
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, x.scalar_type(), "rms_norm_fwd", [&] {
    using T = std::conditional_t<std::is_same_v<scalar_t, at::BFloat16>, __hip_bfloat16,
              std::conditional_t<std::is_same_v<scalar_t, at::Half>, __half, float>>;
    launch_norm_fwd<T>(x, w, bias, y, mean, rstd, eps, true, rows, cols, stream);
  });
  return {y, rstd};
}

std::vector<at::Tensor> rms_norm_bwd(at::Tensor dy, at::Tensor x, at::Tensor w, at::Tensor rstd) {
  int cols = x.size(-1);
  long rows = x.numel() / cols;
  auto dx = at::empty_like(x);
  auto dw = at::zeros({cols}, x.options().dtype(at::kFloat));
  auto db = at::Tensor();
  auto stream = at::hip::getCurrentHIPStream();
  c10::optional<at::Tensor> mean_opt;
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, x.scalar_type(), "rms_norm_bwd", [&] {
    using T = std::conditional_t<std::is_same_v<scalar_t, at::BFloat16>, __hip_bfloat16,
              std::conditional_t<std::is_same_v<scalar_t, at::Half>, __half, float>>;
    auto db_tmp = at::zeros({cols}, x.options().dtype(at::kFloat));
    launch_norm_bwd<T>(dy, x, w, mean_opt, rstd, dx, dw, db_tmp, true, rows, cols, stream);
  });
  return {dx, dw.to(w.scalar_type())};
}
