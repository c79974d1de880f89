// NHWC depthwise 2D convolution (fwd + bwd-data + bwd-weight) for gfx950.
//
// Replaces MIOpen's bf16-NHWC depthwise path, which falls back to
// naive_conv_*_ushort_double_ushort (double-precision accumulation!) and
// makes ConvNeXt training 91% depthwise conv (measured: rocprofv3 on
// convnext_base b128, gpurun_out/prof_cnx).  ConvNeXt-B shapes:
// C in {128..1024}, spatial 56^2..7^2, K=7, stride 1; EfficientNet adds
// K in {3,5}, stride 2.
//
// Layout: x/dy [B,H,W,C] channels-last (C contiguous); weight passed
// pre-transposed as [K*K, C] so per-tap loads vectorize.  Each thread
// owns 8 channels (bf16x8) of one output pixel; fp32 accumulation;
// neighboring-pixel x reuse comes from L1/L2 (weights are L2-resident).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace {

typedef __bf16 bf16x8_t __attribute__((ext_vector_type(8)));

constexpr int kThreads = 256;

// ---- forward: y[b,ho,wo,c] = sum_k x[b, ho*s-p+kh, wo*s-p+kw, c] * w[kh,kw,c] ----
template <typename T>
__global__ __launch_bounds__(kThreads)
void dwconv_fwd_kernel(
    const T* __restrict__ x,      // [B,H,W,C]
    const T* __restrict__ w,      // [K*K, C] (transposed)
    const T* __restrict__ bias,   // [C] or null
    T* __restrict__ y,            // [B,Ho,Wo,C]
    int B, int H, int W, int C,
    int Ho, int Wo, int K, int stride, int pad) {
  const int c8 = C / 8;
  long idx = (long)blockIdx.x * kThreads + threadIdx.x;
  const long total = (long)B * Ho * Wo * c8;
  if (idx >= total) return;

  const int cv = idx % c8;          // channel vector index
  long p = idx / c8;
  const int wo = p % Wo;
  p /= Wo;
  const int ho = p % Ho;
  const int b = p / Ho;
  const int c0 = cv * 8;

  float acc[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) acc[j] = 0.f;

  const int hi0 = ho * stride - pad;
  const int wi0 = wo * stride - pad;
  const T* xb = x + ((long)b * H * W) * C + c0;
  const T* wp = w + c0;
  for (int kh = 0; kh < K; ++kh) {
    const int hi = hi0 + kh;
    if (hi < 0 || hi >= H) continue;
    for (int kw = 0; kw < K; ++kw) {
      const int wi = wi0 + kw;
      if (wi < 0 || wi >= W) continue;
      bf16x8_t xv = *reinterpret_cast<const bf16x8_t*>(xb + ((long)hi * W + wi) * C);
      bf16x8_t wv = *reinterpret_cast<const bf16x8_t*>(wp + (kh * K + kw) * C);
#pragma unroll
      for (int j = 0; j < 8; ++j) acc[j] += (float)xv[j] * (float)wv[j];
    }
  }
  bf16x8_t out;
  if (bias) {
    bf16x8_t bv = *reinterpret_cast<const bf16x8_t*>(bias + c0);
#pragma unroll
    for (int j = 0; j < 8; ++j) out[j] = (__bf16)(acc[j] + (float)bv[j]);
  } else {
#pragma unroll
    for (int j = 0; j < 8; ++j) out[j] = (__bf16)acc[j];
  }
  *reinterpret_cast<bf16x8_t*>(y + (((long)b * Ho + ho) * Wo + wo) * C + c0) = out;
}

// ---- backward data: dx[b,hi,wi,c] = sum_k dy[b,(hi+p-kh)/s,(wi+p-kw)/s,c] * w[kh,kw,c] ----
template <typename T>
__global__ __launch_bounds__(kThreads)
void dwconv_bwd_data_kernel(
    const T* __restrict__ dy,     // [B,Ho,Wo,C]
    const T* __restrict__ w,      // [K*K, C]
    T* __restrict__ dx,           // [B,H,W,C]
    int B, int H, int W, int C,
    int Ho, int Wo, int K, int stride, int pad) {
  const int c8 = C / 8;
  long idx = (long)blockIdx.x * kThreads + threadIdx.x;
  const long total = (long)B * H * W * c8;
  if (idx >= total) return;

  const int cv = idx % c8;
  long p = idx / c8;
  const int wi = p % W;
  p /= W;
  const int hi = p % H;
  const int b = p / H;
  const int c0 = cv * 8;

  float acc[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) acc[j] = 0.f;

  const T* dyb = dy + ((long)b * Ho * Wo) * C + c0;
  const T* wp = w + c0;
  for (int kh = 0; kh < K; ++kh) {
    const int hnum = hi + pad - kh;
    if (hnum < 0 || hnum % stride) continue;
    const int ho = hnum / stride;
    if (ho >= Ho) continue;
    for (int kw = 0; kw < K; ++kw) {
      const int wnum = wi + pad - kw;
      if (wnum < 0 || wnum % stride) continue;
      const int wo = wnum / stride;
      if (wo >= Wo) continue;
      bf16x8_t gv = *reinterpret_cast<const bf16x8_t*>(dyb + ((long)ho * Wo + wo) * C);
      bf16x8_t wv = *reinterpret_cast<const bf16x8_t*>(wp + (kh * K + kw) * C);
#pragma unroll
      for (int j = 0; j < 8; ++j) acc[j] += (float)gv[j] * (float)wv[j];
    }
  }
  bf16x8_t out;
#pragma unroll
  for (int j = 0; j < 8; ++j) out[j] = (__bf16)acc[j];
  *reinterpret_cast<bf16x8_t*>(dx + (((long)b * H + hi) * W + wi) * C + c0) = out;
}

// ---- backward weight: dw[kh,kw,c] = sum_{b,ho,wo} dy[b,ho,wo,c] * x[...] ----
// Grid: (spatial-chunk, kh, channel-superchunk).  Lane layout is
// channel-ACROSS-lanes: lane%16 picks an 8-wide channel chunk, lane/16 picks
// one of 4 pixels, so a wave's dy/x loads are 4 fully-coalesced 256B pixel
// rows instead of 64 scattered 16B strided reads (the v1 layout's ~8x read
// amplification made this kernel 48% of a ConvNeXt train step).  All kw taps
// of one kernel row accumulate in registers; 4-pixel xor-shfl reduce, then
// LDS across waves, then one fp32 atomic per (tap,channel) per block.
template <typename T, int kMaxK>
__global__ __launch_bounds__(kThreads)
void dwconv_bwd_weight_kernel(
    const T* __restrict__ dy,     // [B,Ho,Wo,C]
    const T* __restrict__ x,      // [B,H,W,C]
    float* __restrict__ dw,       // [K*K, C] fp32 (pre-zeroed)
    float* __restrict__ dbias,    // [C] fp32 (pre-zeroed) or null
    int B, int H, int W, int C,
    int Ho, int Wo, int K, int stride, int pad) {
  constexpr int kLanesPerPix = 16;              // channel chunks covered per pixel
  constexpr int kPixPerBlock = kThreads / kLanesPerPix;  // 16
  const int c8 = C / 8;
  const int kh = blockIdx.y;
  const int chunk = blockIdx.z * kLanesPerPix + (threadIdx.x % kLanesPerPix);
  const int pix = threadIdx.x / kLanesPerPix;   // 0..15 within block
  const bool active = chunk < c8;
  const int c0 = chunk * 8;

  const long spatial = (long)B * Ho * Wo;
  float acc[kMaxK][8];
  float bacc[8];
#pragma unroll
  for (int kw = 0; kw < kMaxK; ++kw)
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[kw][j] = 0.f;
#pragma unroll
  for (int j = 0; j < 8; ++j) bacc[j] = 0.f;

  const bool do_bias = (dbias != nullptr) && (kh == 0);
  if (active) {
    for (long s = (long)blockIdx.x * kPixPerBlock + pix; s < spatial;
         s += (long)gridDim.x * kPixPerBlock) {
      const int wo = s % Wo;
      long p = s / Wo;
      const int ho = p % Ho;
      const int b = p / Ho;
      const int hi = ho * stride - pad + kh;
      bf16x8_t gv = *reinterpret_cast<const bf16x8_t*>(
          dy + (((long)b * Ho + ho) * Wo + wo) * C + c0);
      if (do_bias) {
#pragma unroll
        for (int j = 0; j < 8; ++j) bacc[j] += (float)gv[j];
      }
      if (hi < 0 || hi >= H) continue;
      const T* xrow = x + (((long)b * H + hi) * W) * C + c0;
      const int wi0 = wo * stride - pad;
      for (int kw = 0; kw < K; ++kw) {
        const int wi = wi0 + kw;
        if (wi < 0 || wi >= W) continue;
        bf16x8_t xv = *reinterpret_cast<const bf16x8_t*>(xrow + (long)wi * C);
#pragma unroll
        for (int j = 0; j < 8; ++j) acc[kw][j] += (float)gv[j] * (float)xv[j];
      }
    }
  }

  // reduce the 4 pixels of each wave (lanes xor 16, 32 share a channel chunk)
#pragma unroll
  for (int kw = 0; kw < kMaxK; ++kw) {
    if (kw >= K) break;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      acc[kw][j] += __shfl_xor(acc[kw][j], 16, 64);
      acc[kw][j] += __shfl_xor(acc[kw][j], 32, 64);
    }
  }
  if (do_bias) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      bacc[j] += __shfl_xor(bacc[j], 16, 64);
      bacc[j] += __shfl_xor(bacc[j], 32, 64);
    }
  }

  // cross-wave reduce in LDS: [chunk-in-block][tap][8]
  __shared__ float red[kLanesPerPix][kMaxK][8];
  __shared__ float redb[kLanesPerPix][8];
  if (threadIdx.x < kLanesPerPix) {
#pragma unroll
    for (int kw = 0; kw < kMaxK; ++kw)
#pragma unroll
      for (int j = 0; j < 8; ++j) red[threadIdx.x][kw][j] = 0.f;
#pragma unroll
    for (int j = 0; j < 8; ++j) redb[threadIdx.x][j] = 0.f;
  }
  __syncthreads();
  const int lane = threadIdx.x % WAVE_SIZE;
  if (lane < kLanesPerPix && active) {
    const int cib = threadIdx.x % kLanesPerPix;
    for (int kw = 0; kw < K; ++kw) {
#pragma unroll
      for (int j = 0; j < 8; ++j) atomicAdd(&red[cib][kw][j], acc[kw][j]);
    }
    if (do_bias) {
#pragma unroll
      for (int j = 0; j < 8; ++j) atomicAdd(&redb[cib][j], bacc[j]);
    }
  }
  __syncthreads();

  // one global atomic per (tap, channel) per block
  if (threadIdx.x < kLanesPerPix * 8) {
    const int cib = threadIdx.x / 8;
    const int j = threadIdx.x % 8;
    const int c = (blockIdx.z * kLanesPerPix + cib) * 8 + j;
    if (c < C) {
      for (int kw = 0; kw < K; ++kw) {
        atomicAdd(&dw[((long)kh * K + kw) * C + c], red[cib][kw][j]);
      }
      if (do_bias) atomicAdd(&dbias[c], redb[cib][j]);
    }
  }
}

// ---- stride-1 row-sliding variant: each lane walks one output row with a
// K-deep ring of x vectors in registers, so x is loaded ONCE per (row, kh)
// instead of K times (NHWC neighbours are C*2 bytes apart -> no cache-line
// sharing across kw; the generic kernel pays K^2 total read amplification,
// this one pays K).
template <typename T, int kMaxK>
__global__ __launch_bounds__(kThreads)
void dwconv_bwd_weight_s1_kernel(
    const T* __restrict__ dy,     // [B,Ho,Wo,C]
    const T* __restrict__ x,      // [B,H,W,C]
    float* __restrict__ dw,       // [K*K, C] fp32 (pre-zeroed)
    float* __restrict__ dbias,    // [C] fp32 (pre-zeroed) or null
    int B, int H, int W, int C,
    int Ho, int Wo, int pad) {
  constexpr int K = kMaxK;  // compile-time K keeps the ring fully in registers
  constexpr int kLanesPerPix = 16;
  constexpr int kRowsPerBlock = kThreads / kLanesPerPix;  // 16
  const int c8 = C / 8;
  const int kh = blockIdx.y;
  const int chunk = blockIdx.z * kLanesPerPix + (threadIdx.x % kLanesPerPix);
  const int rix = threadIdx.x / kLanesPerPix;
  const bool active = chunk < c8;
  const int c0 = chunk * 8;

  const long rows = (long)B * Ho;
  float acc[kMaxK][8];
  float bacc[8];
#pragma unroll
  for (int kw = 0; kw < kMaxK; ++kw)
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[kw][j] = 0.f;
#pragma unroll
  for (int j = 0; j < 8; ++j) bacc[j] = 0.f;

  const bool do_bias = (dbias != nullptr) && (kh == 0);
  if (active) {
    for (long r = (long)blockIdx.x * kRowsPerBlock + rix; r < rows;
         r += (long)gridDim.x * kRowsPerBlock) {
      const int ho = r % Ho;
      const int b = r / Ho;
      const int hi = ho + kh - pad;   // stride == 1
      const bool row_ok = (hi >= 0 && hi < H);
      const T* dyrow = dy + (((long)b * Ho + ho) * Wo) * C + c0;
      const T* xrow = x + (((long)b * H + (row_ok ? hi : 0)) * W) * C + c0;

      // K-deep window of x vectors; STATIC shifting only — runtime-indexed
      // register arrays spill to scratch and serialize the whole loop
      bf16x8_t ring[kMaxK];
#pragma unroll
      for (int i = 0; i < kMaxK; ++i) ring[i] = bf16x8_t{};
      // preload taps for wo = 0 into slots 0..K-2 (slot i == tap kw=i)
#pragma unroll
      for (int i = 0; i < kMaxK - 1; ++i) {
        if (i >= K - 1) break;
        const int wi = i - pad;
        if (row_ok && wi >= 0 && wi < W)
          ring[i] = *reinterpret_cast<const bf16x8_t*>(xrow + (long)wi * C);
      }
      for (int wo = 0; wo < Wo; ++wo) {
        const int wi_new = wo - pad + K - 1;
        bf16x8_t nv = bf16x8_t{};
        if (row_ok && wi_new >= 0 && wi_new < W)
          nv = *reinterpret_cast<const bf16x8_t*>(xrow + (long)wi_new * C);
        ring[K - 1] = nv;

        bf16x8_t gv = *reinterpret_cast<const bf16x8_t*>(dyrow + (long)wo * C);
        if (do_bias) {
#pragma unroll
          for (int j = 0; j < 8; ++j) bacc[j] += (float)gv[j];
        }
        if (row_ok) {
#pragma unroll
          for (int kw = 0; kw < kMaxK; ++kw) {
            if (kw >= K) break;
#pragma unroll
            for (int j = 0; j < 8; ++j) acc[kw][j] += (float)gv[j] * (float)ring[kw][j];
          }
        }
        // static shift: slot i <- slot i+1 (tap kw at next wo)
#pragma unroll
        for (int i = 0; i < kMaxK - 1; ++i) ring[i] = ring[i + 1];
      }
    }
  }

  // reduce the 4 row-lanes of each wave (lanes xor 16, 32 share a channel chunk)
#pragma unroll
  for (int kw = 0; kw < kMaxK; ++kw) {
    if (kw >= K) break;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      acc[kw][j] += __shfl_xor(acc[kw][j], 16, 64);
      acc[kw][j] += __shfl_xor(acc[kw][j], 32, 64);
    }
  }
  if (do_bias) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      bacc[j] += __shfl_xor(bacc[j], 16, 64);
      bacc[j] += __shfl_xor(bacc[j], 32, 64);
    }
  }

  __shared__ float red[kLanesPerPix][kMaxK][8];
  __shared__ float redb[kLanesPerPix][8];
  if (threadIdx.x < kLanesPerPix) {
#pragma unroll
    for (int kw = 0; kw < kMaxK; ++kw)
#pragma unroll
      for (int j = 0; j < 8; ++j) red[threadIdx.x][kw][j] = 0.f;
#pragma unroll
    for (int j = 0; j < 8; ++j) redb[threadIdx.x][j] = 0.f;
  }
  __syncthreads();
  const int lane = threadIdx.x % WAVE_SIZE;
  if (lane < kLanesPerPix && active) {
    const int cib = threadIdx.x % kLanesPerPix;
    for (int kw = 0; kw < K; ++kw) {
#pragma unroll
      for (int j = 0; j < 8; ++j) atomicAdd(&red[cib][kw][j], acc[kw][j]);
    }
    if (do_bias) {
#pragma unroll
      for (int j = 0; j < 8; ++j) atomicAdd(&redb[cib][j], bacc[j]);
    }
  }
  __syncthreads();

  if (threadIdx.x < kLanesPerPix * 8) {
    const int cib = threadIdx.x / 8;
    const int j = threadIdx.x % 8;
    const int c = (blockIdx.z * kLanesPerPix + cib) * 8 + j;
    if (c < C) {
      for (int kw = 0; kw < K; ++kw) {
        atomicAdd(&dw[((long)kh * K + kw) * C + c], red[cib][kw][j]);
      }
      if (do_bias) atomicAdd(&dbias[c], redb[cib][j]);
    }
  }
}

template <typename scalar_t> struct ToHipD { using type = float; };
template <> struct ToHipD<at::BFloat16> { using type = __hip_bfloat16; };

}  // namespace

at::Tensor dwconv_fwd(at::Tensor x, at::Tensor w_t, c10::optional<at::Tensor> bias,
                      long stride, long pad, long K, long Ho, long Wo) {
  // x: [B,H,W,C] view of channels-last tensor; w_t: [K*K, C] contiguous
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.scalar_type() == at::kBFloat16);
  TORCH_CHECK(x.size(3) % 8 == 0, "dwconv: C must be divisible by 8");
  int B = x.size(0), H = x.size(1), W = x.size(2), C = x.size(3);
  auto y = at::empty({B, Ho, Wo, C}, x.options());
  auto stream = at::hip::getCurrentHIPStream();
  long total = (long)B * Ho * Wo * (C / 8);
  long blocks = (total + kThreads - 1) / kThreads;
  hipLaunchKernelGGL((dwconv_fwd_kernel<__bf16>), dim3(blocks), dim3(kThreads), 0, stream,
      (const __bf16*)x.data_ptr(), (const __bf16*)w_t.data_ptr(),
      bias.has_value() ? (const __bf16*)bias->data_ptr() : nullptr,
      (__bf16*)y.data_ptr(), B, H, W, C, Ho, Wo, (int)K, (int)stride, (int)pad);
  HIP_CHECK_LAST();
  return y;
}

at::Tensor dwconv_bwd_data(at::Tensor dy, at::Tensor w_t, long stride, long pad, long K,
                           long H, long W) {
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous() && dy.scalar_type() == at::kBFloat16);
  int B = dy.size(0), Ho = dy.size(1), Wo = dy.size(2), C = dy.size(3);
  auto dx = at::empty({B, H, W, C}, dy.options());
  auto stream = at::hip::getCurrentHIPStream();
  long total = (long)B * H * W * (C / 8);
  long blocks = (total + kThreads - 1) / kThreads;
  hipLaunchKernelGGL((dwconv_bwd_data_kernel<__bf16>), dim3(blocks), dim3(kThreads), 0, stream,
      (const __bf16*)dy.data_ptr(), (const __bf16*)w_t.data_ptr(),
      (__bf16*)dx.data_ptr(), B, (int)H, (int)W, C, Ho, Wo, (int)K, (int)stride, (int)pad);
  HIP_CHECK_LAST();
  return dx;
}

std::vector<at::Tensor> dwconv_bwd_weight(at::Tensor dy, at::Tensor x, long stride, long pad,
                                          long K, bool need_bias) {
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous() && x.is_contiguous());
  int B = dy.size(0), Ho = dy.size(1), Wo = dy.size(2), C = dy.size(3);
  int H = x.size(1), W = x.size(2);
  auto dw = at::zeros({K * K, C}, dy.options().dtype(at::kFloat));
  at::Tensor dbias;
  float* dbias_ptr = nullptr;
  if (need_bias) {
    dbias = at::zeros({C}, dy.options().dtype(at::kFloat));
    dbias_ptr = dbias.data_ptr<float>();
  }
  auto stream = at::hip::getCurrentHIPStream();
  long spatial = (long)B * Ho * Wo;
  // spatial chunks sized for ~2-4 blocks/CU; atomic depth per dw element is
  // bounded by sblocks (LDS pre-reduction leaves one atomic per block)
  int zc = (int)((C / 8 + 15) / 16);
  long per_pix = (spatial + 15) / 16;
  int sblocks = (int)std::min(std::max<long>(1024 / (K * zc) + 1, 16), per_pix);
  dim3 grid(sblocks, K, zc);
  TORCH_CHECK(K <= 9, "dwconv_bwd_weight: kernel size <= 9 supported");
  auto launch = [&](auto tag) {
    if (stride == 1 && K == decltype(tag)::value) {
      // row-sliding variant: grid.x sized over rows (B*Ho)
      long per_row = ((long)B * Ho + 15) / 16;
      int rb = (int)std::min(std::max<long>(1024 / (K * zc) + 1, 16), per_row);
      dim3 grid_s1(rb, K, zc);
      hipLaunchKernelGGL((dwconv_bwd_weight_s1_kernel<__bf16, decltype(tag)::value>),
          grid_s1, dim3(kThreads), 0, stream,
          (const __bf16*)dy.data_ptr(), (const __bf16*)x.data_ptr(),
          dw.data_ptr<float>(), dbias_ptr, B, H, W, C, Ho, Wo, (int)pad);
      return;
    }
    hipLaunchKernelGGL((dwconv_bwd_weight_kernel<__bf16, decltype(tag)::value>),
        grid, dim3(kThreads), 0, stream,
        (const __bf16*)dy.data_ptr(), (const __bf16*)x.data_ptr(),
        dw.data_ptr<float>(), dbias_ptr, B, H, W, C, Ho, Wo, (int)K, (int)stride, (int)pad);
  };
  if (K <= 3) launch(std::integral_constant<int, 3>{});
  else if (K <= 5) launch(std::integral_constant<int, 5>{});
  else if (K <= 7) launch(std::integral_constant<int, 7>{});
  else launch(std::integral_constant<int, 9>{});
  HIP_CHECK_LAST();
  if (need_bias) return {dw, dbias};
  return {dw};
}
