// Flash-style fused multi-head attention forward for gfx950 (CDNA4).
//
// Replaces the reference's F.scaled_dot_product_attention hot path
// (timm/layers/attention.py:124-129, timm/models/eva.py:246-251).
//
// Structure (correctness-first v1, per the CDNA4 guide's fused-attn anatomy):
//  * 256-thread block = 4 waves; each block owns a 64-row Q tile of one (b,h)
//  * K/V staged in LDS in 32-row tiles; Q staged once per block
//  * QK^T and PV on MFMA v_mfma_f32_16x16x32_bf16 (per-wave 16x16 tiles)
//  * online softmax with fp32 running max/sum state per q-row
//    (C/D fragment layout: row=(lane>>4)*4+reg, col=lane&15 — all lanes of a
//     16-lane group share 4 rows, row-reduce = shfl over 16 lanes)
//  * optional additive fp32 mask [B,1,Nq,Nk] (NaFlex padding masks)
//  * returns O (bf16) + logsumexp (fp32) for the exact GEMM-based backward
//
// Supported: dtype bf16, head_dim in {32,64,96,128}, any Nq/Nk.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace {

typedef __bf16 bf16x8_t __attribute__((ext_vector_type(8)));

constexpr int kQTile = 64;     // q rows per block
constexpr int kKvTile = 64;    // kv rows per LDS stage (4 fragment columns)
constexpr int kBlockThreads = 256;
constexpr float kNegInf = -1e30f;

// LDS XOR swizzle for stride-D row-major bf16 tiles (guide §6 Guideline 4):
// byte_off ^= (row & 7) << 4 spreads the 16 same-column lanes over 8 banks.
template <bool kSwz>
__device__ __forceinline__ int swz(int row, int byte_off) {
  return kSwz ? (byte_off ^ ((row & 7) << 4)) : byte_off;
}

// swizzle for 64-byte LDS rows (transposed V tile over the 32-row kv dim)
template <bool kSwz>
__device__ __forceinline__ int swz64(int row, int byte_off) {
  return kSwz ? (byte_off ^ ((row & 3) << 4)) : byte_off;
}

template <int kMaxD, bool kHasMask, bool kSwizzle>
__global__ __launch_bounds__(kBlockThreads)
void attn_fwd_kernel(
    const __bf16* __restrict__ q,     // [B,H,Nq,D] via strides (last dim contiguous)
    const __bf16* __restrict__ k,
    const __bf16* __restrict__ v,
    const float* __restrict__ mask,   // [B|1,H|1,Nq,Nk] via m_sb/m_sh, or null
    __bf16* __restrict__ o,           // written [B,Nq,H,D] (BNHD) so proj reshape is free
    float* __restrict__ lse,          // [B,H,Nq]
    int B, int H, int Nq, int Nk, int D, float scale,
    // mask batch index is (b % mB): mB==1 broadcasts, mB==B is per-batch, and
    // mB==num_windows serves window-cyclic masks (Swin shift) for free.
    int mB, long m_sb, long m_sh,
    long q_sb, long q_sh, long q_sn,
    long k_sb, long k_sh, long k_sn,
    long v_sb, long v_sh, long v_sn) {
  // LDS: Q[64][D] | K[32][D] | V[32][D] | P[4][16][32]
  __shared__ __bf16 q_lds[kQTile * kMaxD];
  __shared__ __bf16 k_lds[kKvTile * kMaxD];
  __shared__ __bf16 vt_lds[kMaxD * kKvTile];  // V transposed: [d][kv], 64B rows
  __shared__ __bf16 p_lds[4 * 16 * kKvTile];

  const int bh = blockIdx.y;
  const int b = bh / H;
  const int h = bh % H;
  const int qbase = blockIdx.x * kQTile;
  const int wave = threadIdx.x / WAVE_SIZE;
  const int lane = threadIdx.x % WAVE_SIZE;
  const int l16 = lane & 15;      // within-frag row/col index
  const int g4 = lane >> 4;       // 16-lane group id (0..3)

  const __bf16* q_bh = q + (long)b * q_sb + (long)h * q_sh;
  const __bf16* k_bh = k + (long)b * k_sb + (long)h * k_sh;
  const __bf16* v_bh = v + (long)b * v_sb + (long)h * v_sh;
  const int d8 = D / 8;           // 16B chunks per row

  // ---- stage Q tile (cooperative, 16B chunks) ----
  {
    const int chunks = kQTile * d8;
    for (int c = threadIdx.x; c < chunks; c += kBlockThreads) {
      int row = c / d8, col8 = c % d8;
      bf16x8_t val = {};
      if (qbase + row < Nq) {
        val = *reinterpret_cast<const bf16x8_t*>(q_bh + (long)(qbase + row) * q_sn + col8 * 8);
      }
      *reinterpret_cast<bf16x8_t*>(
          reinterpret_cast<char*>(q_lds) + swz<kSwizzle>(row, row * D * 2 + col8 * 16)) = val;
    }
  }
  __syncthreads();

  // ---- per-row online softmax state (4 rows per lane-group, replicated over 16 lanes) ----
  float m_run[4], l_run[4];
  f32x4 acc_o[kMaxD / 16];
#pragma unroll
  for (int r = 0; r < 4; ++r) { m_run[r] = kNegInf; l_run[r] = 0.f; }
#pragma unroll
  for (int f = 0; f < kMaxD / 16; ++f) acc_o[f] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int n_kv_tiles = (Nk + kKvTile - 1) / kKvTile;
  for (int kt = 0; kt < n_kv_tiles; ++kt) {
    const int kv0 = kt * kKvTile;
    // ---- stage K/V tile ----
    {
      const int chunks = kKvTile * d8;
      for (int c = threadIdx.x; c < chunks; c += kBlockThreads) {
        int row = c / d8, col8 = c % d8;
        bf16x8_t kval = {}, vval = {};
        if (kv0 + row < Nk) {
          kval = *reinterpret_cast<const bf16x8_t*>(k_bh + (long)(kv0 + row) * k_sn + col8 * 8);
          vval = *reinterpret_cast<const bf16x8_t*>(v_bh + (long)(kv0 + row) * v_sn + col8 * 8);
        }
        *reinterpret_cast<bf16x8_t*>(
            reinterpret_cast<char*>(k_lds) + swz<kSwizzle>(row, row * D * 2 + col8 * 16)) = kval;
        // V is read column-wise in PV: stage TRANSPOSED so the B-fragment
        // loads below are single ds_read_b128 instead of 8 scalar reads
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          int d = col8 * 8 + j;
          *reinterpret_cast<__bf16*>(
              reinterpret_cast<char*>(vt_lds) +
              swz64<kSwizzle>(d, d * kKvTile * 2 + row * 2)) = vval[j];
        }
      }
    }
    __syncthreads();

    // ---- S = scale * Q K^T (+mask), kKvTile/16 fragments along kv ----
    constexpr int kNF = kKvTile / 16;
    float p_frag[kNF][4];  // [n16][reg] exp'd probabilities
    float m_tile[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) m_tile[r] = kNegInf;

    float s_frag[kNF][4];
#pragma unroll
    for (int n16 = 0; n16 < kNF; ++n16) {
      f32x4 acc = {0.f, 0.f, 0.f, 0.f};
      for (int d0 = 0; d0 < D; d0 += 32) {
        bf16x8_t a = *reinterpret_cast<const bf16x8_t*>(
            reinterpret_cast<const char*>(q_lds) +
            swz<kSwizzle>(wave * 16 + l16, (wave * 16 + l16) * D * 2 + (d0 + g4 * 8) * 2));
        bf16x8_t bfr = *reinterpret_cast<const bf16x8_t*>(
            reinterpret_cast<const char*>(k_lds) +
            swz<kSwizzle>(n16 * 16 + l16, (n16 * 16 + l16) * D * 2 + (d0 + g4 * 8) * 2));
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bfr, acc, 0, 0, 0);
      }
      const int kvcol = kv0 + n16 * 16 + l16;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float s = acc[r] * scale;
        const int qrow = qbase + wave * 16 + g4 * 4 + r;
        if (kvcol >= Nk || qrow >= Nq) {
          s = kNegInf;
        } else if (kHasMask) {
          float mv = mask[(long)(b % mB) * m_sb + (long)h * m_sh + (long)qrow * Nk + kvcol];
          s += mv;
          if (s < kNegInf) s = kNegInf;
        }
        s_frag[n16][r] = s;
        m_tile[r] = fmaxf(m_tile[r], s);
      }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) m_tile[r] = group16_reduce_max(m_tile[r]);

    // ---- online softmax update ----
    float factor[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float m_new = fmaxf(m_run[r], m_tile[r]);
      factor[r] = (m_run[r] == kNegInf) ? 0.f : __expf(m_run[r] - m_new);
      m_run[r] = m_new;
      float psum = 0.f;
#pragma unroll
      for (int n16 = 0; n16 < kNF; ++n16) {
        float p = (s_frag[n16][r] <= kNegInf) ? 0.f : __expf(s_frag[n16][r] - m_new);
        p_frag[n16][r] = p;
        psum += p;
      }
      l_run[r] = l_run[r] * factor[r] + group16_reduce_sum(psum);
    }

    // ---- write P to this wave's LDS region (16 rows x kKvTile cols) ----
#pragma unroll
    for (int n16 = 0; n16 < kNF; ++n16) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        p_lds[wave * (16 * kKvTile) + (g4 * 4 + r) * kKvTile + n16 * 16 + l16] =
            (__bf16)p_frag[n16][r];
      }
    }
    __asm__ volatile("s_waitcnt lgkmcnt(0)" ::: "memory");

    // ---- O += P V, rescaling old accumulator; kv extent covered in
    //      kKvTile/32 MFMA k-steps ----
    bf16x8_t pa[kKvTile / 32];
#pragma unroll
    for (int ks = 0; ks < kKvTile / 32; ++ks) {
      pa[ks] = *reinterpret_cast<const bf16x8_t*>(
          p_lds + wave * (16 * kKvTile) + l16 * kKvTile + ks * 32 + g4 * 8);
    }
#pragma unroll
    for (int f = 0; f < kMaxD / 16; ++f) {
      if (f * 16 >= D) break;
#pragma unroll
      for (int r = 0; r < 4; ++r) acc_o[f][r] *= factor[r];
      const int drow = f * 16 + l16;
#pragma unroll
      for (int ks = 0; ks < kKvTile / 32; ++ks) {
        // B operand: V^T[dcol][kv=ks*32+g4*8 .. +8) — one vector LDS read
        bf16x8_t bv = *reinterpret_cast<const bf16x8_t*>(
            reinterpret_cast<const char*>(vt_lds) +
            swz64<kSwizzle>(drow, drow * kKvTile * 2 + (ks * 32 + g4 * 8) * 2));
        acc_o[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pa[ks], bv, acc_o[f], 0, 0, 0);
      }
    }
    __syncthreads();  // before next tile overwrites K/V
  }

  // ---- normalize + write O (BNHD layout), LSE ----
  __bf16* o_bh = o + ((long)b * Nq) * (H * D) + (long)h * D;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int qrow = qbase + wave * 16 + g4 * 4 + r;
    if (qrow >= Nq) continue;
    const float inv_l = (l_run[r] > 0.f) ? 1.f / l_run[r] : 0.f;
    for (int f = 0; f < kMaxD / 16; ++f) {
      if (f * 16 >= D) break;
      o_bh[(long)qrow * (H * D) + f * 16 + l16] = (__bf16)(acc_o[f][r] * inv_l);
    }
    if (l16 == 0) {
      lse[(long)bh * Nq + qrow] =
          (l_run[r] > 0.f) ? (m_run[r] + __logf(l_run[r])) : INFINITY;
    }
  }
}

template <int kMaxD>
void launch_attn_fwd(const at::Tensor& q, const at::Tensor& k, const at::Tensor& v,
                     const c10::optional<at::Tensor>& mask, at::Tensor& o, at::Tensor& lse,
                     int B, int H, int Nq, int Nk, int D, float scale, hipStream_t stream) {
  dim3 grid(cdiv(Nq, kQTile), B * H);
  dim3 block(kBlockThreads);
  const bool has_mask = mask.has_value();
  const float* mp = has_mask ? mask->data_ptr<float>() : nullptr;
  int mB = 1;
  long m_sb = 0, m_sh = 0;
  if (has_mask) {
    mB = (int)mask->size(0);
    m_sb = mB == 1 ? 0 : mask->size(1) * (long)Nq * Nk;
    m_sh = mask->size(1) == 1 ? 0 : (long)Nq * Nk;
  }
  auto args = [&](auto kern) {
    hipLaunchKernelGGL(kern, grid, block, 0, stream,
        (const __bf16*)q.data_ptr(), (const __bf16*)k.data_ptr(), (const __bf16*)v.data_ptr(),
        mp, (__bf16*)o.data_ptr(), lse.data_ptr<float>(), B, H, Nq, Nk, D, scale,
        mB, m_sb, m_sh,
        q.stride(0), q.stride(1), q.stride(2),
        k.stride(0), k.stride(1), k.stride(2),
        v.stride(0), v.stride(1), v.stride(2));
  };
  // XOR swizzle is only bijective within a row when the row stride (D*2 bytes)
  // is a power of two >= 128B; D=32/96 run unswizzled.
  const bool swizzle = (D == 64 || D == 128);
  if (has_mask && swizzle) args(attn_fwd_kernel<kMaxD, true, true>);
  else if (has_mask) args(attn_fwd_kernel<kMaxD, true, false>);
  else if (swizzle) args(attn_fwd_kernel<kMaxD, false, true>);
  else args(attn_fwd_kernel<kMaxD, false, false>);
  HIP_CHECK_LAST();
}

}  // namespace

namespace {

// ---- backward helper kernels: fuse the softmax recompute elementwise ----
// One wave per row (lse/mask shared per row); vectorized when Nk % 8 == 0,
// scalar tail otherwise.
// p = exp(s * scale + mask - lse[row])   (s bf16 in, p bf16 out, fp32 math)
template <bool kHasMask>
__global__ __launch_bounds__(256)
void attn_bwd_softmax_kernel(
    const __bf16* __restrict__ s,    // [B,H,Nq,Nk]
    const float* __restrict__ lse,   // [B,H,Nq]
    const float* __restrict__ mask,  // [B|1,H|1,Nq,Nk] via m_sb/m_sh, or null
    __bf16* __restrict__ p,
    long rows, int Nk, int Nq, int H, float scale,
    int mB, long m_sb, long m_sh) {
  const int wave = threadIdx.x / WAVE_SIZE;
  const int lane = threadIdx.x % WAVE_SIZE;
  constexpr int kWaves = 256 / WAVE_SIZE;
  for (long row = (long)blockIdx.x * kWaves + wave; row < rows;
       row += (long)gridDim.x * kWaves) {
    const float l = lse[row];
    const __bf16* sr = s + row * Nk;
    __bf16* pr = p + row * Nk;
    const float* mrow = nullptr;
    if (kHasMask) {
      long q = row % Nq;
      long h = (row / Nq) % H;
      long b = row / ((long)H * Nq);
      mrow = mask + (b % mB) * m_sb + h * m_sh + q * Nk;
    }
    int nvec = (Nk % 8 == 0) ? Nk / 8 : 0;
    for (int i = lane; i < nvec; i += WAVE_SIZE) {
      bf16x8_t sv = *reinterpret_cast<const bf16x8_t*>(sr + i * 8);
      bf16x8_t out;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = (float)sv[j] * scale;
        if (kHasMask) f += mrow[i * 8 + j];
        out[j] = (__bf16)__expf(f - l);
      }
      *reinterpret_cast<bf16x8_t*>(pr + i * 8) = out;
    }
    for (int i = nvec * 8 + lane; i < Nk; i += WAVE_SIZE) {
      float f = (float)sr[i] * scale;
      if (kHasMask) f += mrow[i];
      pr[i] = (__bf16)__expf(f - l);
    }
  }
}

// ds = p * (dp - delta[row]) * scale   (all bf16 except delta fp32)
__global__ __launch_bounds__(256)
void attn_bwd_ds_kernel(
    const __bf16* __restrict__ p,
    const __bf16* __restrict__ dp,
    const float* __restrict__ delta,  // [B,H,Nq]
    __bf16* __restrict__ ds,
    long rows, int Nk, float scale) {
  const int wave = threadIdx.x / WAVE_SIZE;
  const int lane = threadIdx.x % WAVE_SIZE;
  constexpr int kWaves = 256 / WAVE_SIZE;
  for (long row = (long)blockIdx.x * kWaves + wave; row < rows;
       row += (long)gridDim.x * kWaves) {
    const float d = delta[row];
    const __bf16* pr = p + row * Nk;
    const __bf16* dpr = dp + row * Nk;
    __bf16* dsr = ds + row * Nk;
    int nvec = (Nk % 8 == 0) ? Nk / 8 : 0;
    for (int i = lane; i < nvec; i += WAVE_SIZE) {
      bf16x8_t pv = *reinterpret_cast<const bf16x8_t*>(pr + i * 8);
      bf16x8_t dpv = *reinterpret_cast<const bf16x8_t*>(dpr + i * 8);
      bf16x8_t out;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        out[j] = (__bf16)(((float)pv[j] * ((float)dpv[j] - d)) * scale);
      }
      *reinterpret_cast<bf16x8_t*>(dsr + i * 8) = out;
    }
    for (int i = nvec * 8 + lane; i < Nk; i += WAVE_SIZE) {
      dsr[i] = (__bf16)(((float)pr[i] * ((float)dpr[i] - d)) * scale);
    }
  }
}

}  // namespace

at::Tensor attn_bwd_softmax(at::Tensor s, at::Tensor lse, c10::optional<at::Tensor> mask,
                            double scale) {
  TORCH_CHECK(s.is_cuda() && s.is_contiguous() && s.scalar_type() == at::kBFloat16);
  int H = s.size(1), Nq = s.size(2), Nk = s.size(3);
  auto p = at::empty_like(s);
  long rows = s.numel() / Nk;
  auto stream = at::hip::getCurrentHIPStream();
  int blocks = (int)std::min((long)4096, (rows + 3) / 4);
  if (mask.has_value()) {
    int mB = (int)mask->size(0);
    long m_sb = mB == 1 ? 0 : mask->size(1) * (long)Nq * Nk;
    long m_sh = mask->size(1) == 1 ? 0 : (long)Nq * Nk;
    hipLaunchKernelGGL((attn_bwd_softmax_kernel<true>), dim3(blocks), dim3(256), 0, stream,
        (const __bf16*)s.data_ptr(), lse.data_ptr<float>(), mask->data_ptr<float>(),
        (__bf16*)p.data_ptr(), rows, Nk, Nq, H, (float)scale, mB, m_sb, m_sh);
  } else {
    hipLaunchKernelGGL((attn_bwd_softmax_kernel<false>), dim3(blocks), dim3(256), 0, stream,
        (const __bf16*)s.data_ptr(), lse.data_ptr<float>(), nullptr,
        (__bf16*)p.data_ptr(), rows, Nk, Nq, H, (float)scale, 1, 0, 0);
  }
  HIP_CHECK_LAST();
  return p;
}

at::Tensor attn_bwd_ds(at::Tensor p, at::Tensor dp, at::Tensor delta, double scale) {
  TORCH_CHECK(p.is_cuda() && p.is_contiguous() && dp.is_contiguous());
  int Nk = p.size(3);
  auto ds = at::empty_like(p);
  long rows = p.numel() / Nk;
  auto stream = at::hip::getCurrentHIPStream();
  int blocks = (int)std::min((long)4096, (rows + 3) / 4);
  hipLaunchKernelGGL(attn_bwd_ds_kernel, dim3(blocks), dim3(256), 0, stream,
      (const __bf16*)p.data_ptr(), (const __bf16*)dp.data_ptr(), delta.data_ptr<float>(),
      (__bf16*)ds.data_ptr(), rows, Nk, (float)scale);
  HIP_CHECK_LAST();
  return ds;
}

std::vector<at::Tensor> attention_fwd(at::Tensor q, at::Tensor k, at::Tensor v,
                                      c10::optional<at::Tensor> mask, double scale) {
  TORCH_CHECK(q.is_cuda() && q.stride(3) == 1 && k.stride(3) == 1 && v.stride(3) == 1,
              "attention_fwd: innermost (head_dim) stride must be 1");
  TORCH_CHECK(q.scalar_type() == at::kBFloat16, "attention_fwd: bf16 only");
  TORCH_CHECK(q.dim() == 4);
  int B = q.size(0), H = q.size(1), Nq = q.size(2), D = q.size(3);
  int Nk = k.size(2);
  TORCH_CHECK(D % 32 == 0 && D <= 128, "attention_fwd: head_dim must be multiple of 32, <=128");
  if (mask.has_value()) {
    TORCH_CHECK(mask->is_contiguous() && mask->scalar_type() == at::kFloat);
    TORCH_CHECK(mask->size(0) >= 1 && B % mask->size(0) == 0 &&
                (mask->size(1) == H || mask->size(1) == 1) &&
                mask->size(2) == Nq && mask->size(3) == Nk,
                "attention_fwd: mask must be [mB|1,H|1,Nq,Nk] with mB dividing B");
  }
  // O allocated [B, Nq, H, D] and returned as a permuted [B,H,Nq,D] view so
  // the caller's transpose(1,2).reshape(B,N,C) is a zero-copy reshape.
  auto o_bnhd = at::empty({B, Nq, H, D}, q.options());
  auto o = o_bnhd.permute({0, 2, 1, 3});
  auto lse = at::empty({B, H, Nq}, q.options().dtype(at::kFloat));
  auto stream = at::hip::getCurrentHIPStream();
  if (D <= 32) launch_attn_fwd<32>(q, k, v, mask, o, lse, B, H, Nq, Nk, D, (float)scale, stream);
  else if (D <= 64) launch_attn_fwd<64>(q, k, v, mask, o, lse, B, H, Nq, Nk, D, (float)scale, stream);
  else if (D <= 96) launch_attn_fwd<96>(q, k, v, mask, o, lse, B, H, Nq, Nk, D, (float)scale, stream);
  else launch_attn_fwd<128>(q, k, v, mask, o, lse, B, H, Nq, Nk, D, (float)scale, stream);
  return {o, lse};
}

namespace {

// ---- bwd preprocess: fuse the do-contiguous copy with delta = rowsum(do*o) ----
// do/o arrive as strided [B,H,N,D] views of BNHD storage (the fwd's output
// layout); one pass reads both, emits contiguous bf16 dO plus fp32 delta.
// Replaces three torch kernels (direct_copy + mul + fp32 reduce) that moved
// ~5x the bytes.  Layout: lanes_per_row = D/8 lanes cooperate on one row
// with bf16x8 vector loads; a 256-thread block covers 256*8/D rows.
__global__ __launch_bounds__(256)
void attn_bwd_preprocess_kernel(
    const __bf16* __restrict__ dov,  // strided [B,H,N,D]
    const __bf16* __restrict__ ov,   // strided [B,H,N,D]
    __bf16* __restrict__ do_c,       // contiguous [B,H,N,D]
    float* __restrict__ delta,       // [B,H,N]
    long rows, int D, int lpr,       // lpr: pow2 lanes per row (host picks)
    long do_sb, long do_sh, long do_sn,
    long o_sb, long o_sh, long o_sn,
    int H, int N) {
  const int vecs = (D / 8) / lpr;              // bf16x8 chunks per lane
  const int rows_per_block = 256 / lpr;
  const int local_row = threadIdx.x / lpr;
  const int lane_in_row = threadIdx.x % lpr;

  for (long row = (long)blockIdx.x * rows_per_block + local_row; row < rows;
       row += (long)gridDim.x * rows_per_block) {
    const long n = row % N;
    const long h = (row / N) % H;
    const long b = row / ((long)H * N);
    const __bf16* do_row = dov + b * do_sb + h * do_sh + n * do_sn;
    const __bf16* o_row = ov + b * o_sb + h * o_sh + n * o_sn;

    float acc = 0.f;
    for (int v = 0; v < vecs; ++v) {
      const int c8 = v * lpr + lane_in_row;
      bf16x8_t dv = *reinterpret_cast<const bf16x8_t*>(do_row + c8 * 8);
      bf16x8_t oval = *reinterpret_cast<const bf16x8_t*>(o_row + c8 * 8);
#pragma unroll
      for (int j = 0; j < 8; ++j) acc += (float)dv[j] * (float)oval[j];
      *reinterpret_cast<bf16x8_t*>(do_c + row * D + c8 * 8) = dv;
    }

    // xor ladder over the lpr (always pow2) lanes of this row
    for (int off = lpr / 2; off > 0; off /= 2) {
      acc += __shfl_xor(acc, off, 64);
    }
    if (lane_in_row == 0) delta[row] = acc;
  }
}

}  // namespace

std::vector<at::Tensor> attn_bwd_preprocess(at::Tensor dout, at::Tensor o) {
  TORCH_CHECK(dout.is_cuda() && dout.dim() == 4 && dout.scalar_type() == at::kBFloat16);
  TORCH_CHECK(dout.stride(3) == 1 && o.stride(3) == 1, "head_dim must be contiguous");
  int B = dout.size(0), H = dout.size(1), N = dout.size(2), D = dout.size(3);
  TORCH_CHECK(D % 32 == 0 && D <= 128);
  auto do_c = at::empty({B, H, N, D}, dout.options());
  auto delta = at::empty({B, H, N}, dout.options().dtype(at::kFloat));
  long rows = (long)B * H * N;
  int lpr = D / 8;
  if (lpr & (lpr - 1)) lpr = 4;  // non-pow2 chunk count (D=96): 4 lanes x 3 chunks
  int rows_per_block = 256 / lpr;
  int blocks = (int)std::min((rows + rows_per_block - 1) / rows_per_block, (long)8192);
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(attn_bwd_preprocess_kernel, dim3(blocks), dim3(256), 0, stream,
      (const __bf16*)dout.data_ptr(), (const __bf16*)o.data_ptr(),
      (__bf16*)do_c.data_ptr(), delta.data_ptr<float>(),
      rows, D, lpr,
      dout.stride(0), dout.stride(1), dout.stride(2),
      o.stride(0), o.stride(1), o.stride(2),
      H, N);
  HIP_CHECK_LAST();
  return {do_c, delta};
}
