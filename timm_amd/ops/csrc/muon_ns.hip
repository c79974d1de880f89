// Batched bf16 MFMA GEMM kernels for the Muon Newton-Schulz iteration
// (gfx950).  Replaces the hipBLASLt torch.matmul path flagged in round 1.
//
// The quintic NS step  A = X X^T;  B = b A + c A^2;  X = a X + B X
// maps to three launches of two kernel shapes (A and B are symmetric, so
// A^2 == A A^T keeps every operand row-major):
//   ns_gemm_nt:  out = alpha * L R^T + beta * S     (L[B,M,K], R[B,N,K])
//   ns_gemm_nn:  out = alpha * L R   + beta * S     (L[B,M,K], R[B,K,N])
//
// Geometry: 128x128 output tile per 256-thread block (4 waves, one 64x64
// quadrant each, 4x4 fragments of v_mfma_f32_16x16x32_bf16), K staged in
// 32-wide LDS tiles with a 4-chunk XOR swizzle; fp32 accumulation, bf16 IO.
// Batch rides grid.z so same-shape parameter groups fill all 256 CUs.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace {

typedef __bf16 bf16x8_t __attribute__((ext_vector_type(8)));

constexpr int kTile = 128;   // BM == BN
constexpr int kK = 32;       // K step
constexpr int kThreads = 256;

// 64B rows (32 bf16): XOR the 16B-chunk index with (row & 3)
__device__ __forceinline__ int swz64(int row, int byte_off) {
  return byte_off ^ ((row & 3) << 4);
}

// Stage a [kTile x kK] operand tile from row-major src (rows along dim of
// length `rows_limit`, k along contiguous dim of length `k_limit`).
__device__ __forceinline__ void stage_rows(
    __bf16* lds, const __bf16* src, long ld, int row0, int rows_limit,
    int k0, int k_limit) {
  // 128 rows x 4 chunks of 16B = 512 chunks; 256 threads -> 2 each
  for (int c = threadIdx.x; c < kTile * (kK / 8); c += kThreads) {
    const int row = c / (kK / 8);
    const int chunk = c % (kK / 8);
    bf16x8_t val = {};
    const int gr = row0 + row;
    const int gk = k0 + chunk * 8;
    if (gr < rows_limit) {
      if (gk + 8 <= k_limit) {
        val = *reinterpret_cast<const bf16x8_t*>(src + (long)gr * ld + gk);
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          val[j] = (gk + j < k_limit) ? src[(long)gr * ld + gk + j] : (__bf16)0.f;
        }
      }
    }
    *reinterpret_cast<bf16x8_t*>(
        reinterpret_cast<char*>(lds) + swz64(row, row * kK * 2 + chunk * 16)) = val;
  }
}

// Stage a [kK x kTile] slab of row-major R (k rows, n contiguous) TRANSPOSED
// into the same [kTile][kK] LDS layout (element [n][k]).
__device__ __forceinline__ void stage_cols_t(
    __bf16* lds, const __bf16* src, long ld, int k0, int k_limit,
    int n0, int n_limit) {
  for (int c = threadIdx.x; c < kK * (kTile / 8); c += kThreads) {
    const int k = c / (kTile / 8);
    const int chunk = c % (kTile / 8);
    bf16x8_t val = {};
    const int gk = k0 + k;
    const int gn = n0 + chunk * 8;
    if (gk < k_limit) {
      if (gn + 8 <= n_limit) {
        val = *reinterpret_cast<const bf16x8_t*>(src + (long)gk * ld + gn);
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          val[j] = (gn + j < n_limit) ? src[(long)gk * ld + gn + j] : (__bf16)0.f;
        }
      }
    }
    // scatter the 8 n-values to their transposed rows
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int row = chunk * 8 + j;  // n index
      *reinterpret_cast<__bf16*>(
          reinterpret_cast<char*>(lds) + swz64(row, row * kK * 2 + k * 2)) = val[j];
    }
  }
}

template <bool kTransB, bool kHasS>
__global__ __launch_bounds__(kThreads)
void ns_gemm_kernel(
    const __bf16* __restrict__ l,   // [B, M, K]
    const __bf16* __restrict__ r,   // NT: [B, N, K]; NN: [B, K, N]
    const __bf16* __restrict__ s,   // [B, M, N] or null
    __bf16* __restrict__ out,       // [B, M, N]
    int M, int N, int K, float alpha, float beta) {
  __shared__ __bf16 l_lds[kTile * kK];
  __shared__ __bf16 r_lds[kTile * kK];

  const int batch = blockIdx.z;
  const int n0 = blockIdx.x * kTile;
  const int m0 = blockIdx.y * kTile;
  const int wave = threadIdx.x / WAVE_SIZE;
  const int lane = threadIdx.x % WAVE_SIZE;
  const int l16 = lane & 15;
  const int g4 = lane >> 4;
  const int wr = (wave >> 1) * 64;  // wave quadrant row offset in tile
  const int wc = (wave & 1) * 64;

  const __bf16* l_b = l + (long)batch * M * K;
  const __bf16* r_b = r + (long)batch * (kTransB ? (long)K * N : (long)N * K);

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  for (int k0 = 0; k0 < K; k0 += kK) {
    stage_rows(l_lds, l_b, K, m0, M, k0, K);
    if (kTransB) {
      stage_cols_t(r_lds, r_b, N, k0, K, n0, N);
    } else {
      stage_rows(r_lds, r_b, K, n0, N, k0, K);
    }
    __syncthreads();

    bf16x8_t a_frag[4], b_frag[4];
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int row = wr + i * 16 + l16;
      a_frag[i] = *reinterpret_cast<const bf16x8_t*>(
          reinterpret_cast<const char*>(l_lds) + swz64(row, row * kK * 2 + g4 * 16));
    }
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int row = wc + j * 16 + l16;
      b_frag[j] = *reinterpret_cast<const bf16x8_t*>(
          reinterpret_cast<const char*>(r_lds) + swz64(row, row * kK * 2 + g4 * 16));
    }
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j)
        acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a_frag[i], b_frag[j], acc[i][j], 0, 0, 0);
    __syncthreads();
  }

  // epilogue: out = alpha*acc + beta*S, bounds-guarded
  const long out_base = (long)batch * M * N;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
#pragma unroll
    for (int rr = 0; rr < 4; ++rr) {
      const int gm = m0 + wr + i * 16 + g4 * 4 + rr;
      if (gm >= M) continue;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int gn = n0 + wc + j * 16 + l16;
        if (gn >= N) continue;
        float v = alpha * acc[i][j][rr];
        if (kHasS) v += beta * (float)s[out_base + (long)gm * N + gn];
        out[out_base + (long)gm * N + gn] = (__bf16)v;
      }
    }
  }
}

void launch_ns_gemm(
    const at::Tensor& l, const at::Tensor& r, const c10::optional<at::Tensor>& s,
    at::Tensor& out, bool trans_b, int B, int M, int N, int K,
    float alpha, float beta, hipStream_t stream) {
  dim3 grid(cdiv(N, kTile), cdiv(M, kTile), B);
  dim3 block(kThreads);
  const bool has_s = s.has_value();
  const __bf16* sp = has_s ? (const __bf16*)s->data_ptr() : nullptr;
  auto args = [&](auto kern) {
    hipLaunchKernelGGL(kern, grid, block, 0, stream,
        (const __bf16*)l.data_ptr(), (const __bf16*)r.data_ptr(), sp,
        (__bf16*)out.data_ptr(), M, N, K, alpha, beta);
  };
  if (trans_b && has_s) args(ns_gemm_kernel<true, true>);
  else if (trans_b) args(ns_gemm_kernel<true, false>);
  else if (has_s) args(ns_gemm_kernel<false, true>);
  else args(ns_gemm_kernel<false, false>);
  HIP_CHECK_LAST();
}

at::Tensor ns_gemm_common(
    at::Tensor l, at::Tensor r, c10::optional<at::Tensor> s,
    double alpha, double beta, bool nt) {
  TORCH_CHECK(l.is_cuda() && l.dim() == 3 && l.is_contiguous() &&
              l.scalar_type() == at::kBFloat16, "ns_gemm: L must be [B,M,K] bf16 contiguous");
  TORCH_CHECK(r.is_contiguous() && r.dim() == 3 && r.scalar_type() == at::kBFloat16);
  int B = l.size(0), M = l.size(1), K = l.size(2);
  int N = nt ? r.size(1) : r.size(2);
  if (nt) {
    TORCH_CHECK(r.size(0) == B && r.size(2) == K, "ns_gemm_nt: R must be [B,N,K]");
  } else {
    TORCH_CHECK(r.size(0) == B && r.size(1) == K, "ns_gemm_nn: R must be [B,K,N]");
  }
  if (s.has_value()) {
    TORCH_CHECK(s->is_contiguous() && s->sizes() == at::IntArrayRef({B, M, N}));
  }
  auto out = at::empty({B, M, N}, l.options());
  auto stream = at::hip::getCurrentHIPStream();
  // kTransB template flag means "stage R transposed" which is the NN case
  launch_ns_gemm(l, r, s, out, !nt, B, M, N, K, (float)alpha, (float)beta, stream);
  return out;
}

}  // namespace

// out = alpha * L @ R^T + beta * S     L:[B,M,K]  R:[B,N,K]
at::Tensor ns_gemm_nt(at::Tensor l, at::Tensor r, c10::optional<at::Tensor> s,
                      double alpha, double beta) {
  return ns_gemm_common(l, r, s, alpha, beta, /*nt=*/true);
}

// out = alpha * L @ R + beta * S       L:[B,M,K]  R:[B,K,N]
at::Tensor ns_gemm_nn(at::Tensor l, at::Tensor r, c10::optional<at::Tensor> s,
                      double alpha, double beta) {
  return ns_gemm_common(l, r, s, alpha, beta, /*nt=*/false);
}
