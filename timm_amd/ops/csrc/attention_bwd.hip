// Fused flash-style attention BACKWARD for gfx950 (CDNA4).
//
// Replaces the round-1 GEMM-recompute backward (which materialized S, P, dP,
// dS as [B,H,Nq,Nk] bf16 tensors in HBM via hipBLASLt).  Reference semantics:
// grad of F.scaled_dot_product_attention (timm/layers/attention.py:124-129).
//
// Two passes, both flash-style (no N^2 HBM tensors, softmax recomputed from
// the forward's saved logsumexp inside the tile loop):
//
//  * attn_bwd_dq_kernel:   grid over (Q-tiles, B*H).  Each 256-thread block
//    owns 64 q rows; loops over 32-row K/V tiles computing
//      S = QK^T -> P = exp(S*scale + mask - lse) -> dP = dO V^T
//      dS = P o (dP - delta) * scale -> dQ += dS K
//    dQ accumulates in fp32 MFMA accumulators (never leaves registers).
//
//  * attn_bwd_dkdv_kernel: grid over (KV-tiles, B*H).  Each block owns 64 kv
//    rows; loops over 32-row Q/dO tiles computing the transposed quantities
//      S^T = K Q^T -> P^T -> dP^T = V dO^T -> dS^T
//      dV += P^T dO,  dK += dS^T Q
//
// All matmuls are MFMA v_mfma_f32_16x16x32_bf16.  LDS tiles use the same XOR
// swizzle as the forward (bank-conflict-free ds_read_b128 row reads); the
// column-wise B-operand gathers apply the same XOR per-element so one staged
// copy serves both access patterns.
//
// dQ/dK/dV outputs are written through explicit strides so they can target a
// packed [B,N,3,H,D] dqkv buffer directly — this removes the qkv-unbind
// `aten::copy_` grad stack (VERDICT round-1, weak #8).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace {

typedef __bf16 bf16x8_t __attribute__((ext_vector_type(8)));

constexpr int kBlockThreads = 256;
constexpr float kNegInf = -1e30f;

template <bool kSwz>
__device__ __forceinline__ int swz(int row, int byte_off) {
  return kSwz ? (byte_off ^ ((row & 7) << 4)) : byte_off;
}

// swizzle for 64-byte LDS rows (transposed tiles over a 32-row dimension)
template <bool kSwz>
__device__ __forceinline__ int swz64(int row, int byte_off) {
  return kSwz ? (byte_off ^ ((row & 3) << 4)) : byte_off;
}

// Cooperative stage of a [rows x D] bf16 tile into (optionally swizzled) LDS,
// zero-filling rows past `n_limit`.  16B chunks, whole block participates.
template <bool kSwz>
__device__ __forceinline__ void stage_tile(
    __bf16* lds, const __bf16* src, long src_sn, int base, int rows, int n_limit,
    int D, int d8) {
  const int chunks = rows * d8;
  for (int c = threadIdx.x; c < chunks; c += kBlockThreads) {
    int row = c / d8, col8 = c % d8;
    bf16x8_t val = {};
    if (base + row < n_limit) {
      val = *reinterpret_cast<const bf16x8_t*>(src + (long)(base + row) * src_sn + col8 * 8);
    }
    *reinterpret_cast<bf16x8_t*>(
        reinterpret_cast<char*>(lds) + swz<kSwz>(row, row * D * 2 + col8 * 16)) = val;
  }
}

// Read one MFMA A/B row fragment (row-major operand) from swizzled LDS:
// lane holds elements [d0 + g4*8 .. +8) of `row`.
template <bool kSwz>
__device__ __forceinline__ bf16x8_t frag_row(const __bf16* lds, int row, int d0, int g4, int D) {
  return *reinterpret_cast<const bf16x8_t*>(
      reinterpret_cast<const char*>(lds) + swz<kSwz>(row, row * D * 2 + (d0 + g4 * 8) * 2));
}

// Gather one MFMA B column fragment: lane holds elements
// T[k = g4*8+j][col] for j=0..7 (scalar reads, swizzle-aware addressing).
// Fallback path for kMaxD > 64 where the transposed tiles don't fit in LDS.
template <bool kSwz>
__device__ __forceinline__ bf16x8_t frag_col(const __bf16* lds, int g4, int col, int D) {
  bf16x8_t out;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    int row = g4 * 8 + j;
    out[j] = *reinterpret_cast<const __bf16*>(
        reinterpret_cast<const char*>(lds) + swz<kSwz>(row, row * D * 2 + col * 2));
  }
  return out;
}

// Cooperative transposed stage: scatter a [rows x D] global tile into
// LDS as T[D][rows] (64-byte rows, swz64). One b128 global read per chunk,
// eight u16 scattered LDS writes — paid once per tile so the MFMA loop can
// read column fragments as vector ds_read_b128 instead of 8 scalar reads.
template <bool kSwz>
__device__ __forceinline__ void stage_tile_t(
    __bf16* lds_t, const __bf16* src, long src_sn, int base, int rows, int n_limit,
    int D, int d8) {
  const int chunks = rows * d8;
  for (int c = threadIdx.x; c < chunks; c += kBlockThreads) {
    int row = c / d8, col8 = c % d8;
    bf16x8_t val = {};
    if (base + row < n_limit) {
      val = *reinterpret_cast<const bf16x8_t*>(src + (long)(base + row) * src_sn + col8 * 8);
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      int d = col8 * 8 + j;
      *reinterpret_cast<__bf16*>(
          reinterpret_cast<char*>(lds_t) + swz64<kSwz>(d, d * rows * 2 + row * 2)) = val[j];
    }
  }
}

// Vector read of a transposed-tile B fragment: lane (l16, g4) takes
// T[drow][g4*8 .. +8) — contiguous, one ds_read_b128.
template <bool kSwz>
__device__ __forceinline__ bf16x8_t frag_row_t(const __bf16* lds_t, int drow, int g4, int rows) {
  return *reinterpret_cast<const bf16x8_t*>(
      reinterpret_cast<const char*>(lds_t) + swz64<kSwz>(drow, drow * rows * 2 + g4 * 8 * 2));
}

// ---------------------------------------------------------------------------
// dQ pass: block owns 64 q rows (wave w -> rows [w*16, w*16+16)), loops kv.
// ---------------------------------------------------------------------------
template <int kMaxD, bool kHasMask, bool kSwizzle>
__global__ __launch_bounds__(kBlockThreads)
void attn_bwd_dq_kernel(
    const __bf16* __restrict__ q,     // [B,H,Nq,D] strided
    const __bf16* __restrict__ k,     // [B,H,Nk,D] strided
    const __bf16* __restrict__ v,     // [B,H,Nk,D] strided
    const __bf16* __restrict__ dov,   // [B,H,Nq,D] contiguous
    const float* __restrict__ lse,    // [B,H,Nq]
    const float* __restrict__ delta,  // [B,H,Nq]
    const float* __restrict__ mask,   // [mB,H|1,Nq,Nk] or null
    __bf16* __restrict__ dq,          // strided out
    int B, int H, int Nq, int Nk, int D, float scale,
    int mB, long m_sb, long m_sh,
    long q_sb, long q_sh, long q_sn,
    long k_sb, long k_sh, long k_sn,
    long v_sb, long v_sh, long v_sn,
    long dq_sb, long dq_sh, long dq_sn) {
  constexpr int kQB = 64, kKvB = 32;
  constexpr bool kUseT = kMaxD <= 64;  // transposed K tile fits in LDS
  __shared__ __bf16 q_lds[kQB * kMaxD];
  __shared__ __bf16 do_lds[kQB * kMaxD];
  __shared__ __bf16 k_lds[kKvB * kMaxD];
  __shared__ __bf16 v_lds[kKvB * kMaxD];
  __shared__ __bf16 ds_lds[4 * 16 * kKvB];
  __shared__ __bf16 kt_lds[kUseT ? kMaxD * kKvB : 1];

  const int bh = blockIdx.y;
  const int b = bh / H;
  const int h = bh % H;
  const int qbase = blockIdx.x * kQB;
  const int wave = threadIdx.x / WAVE_SIZE;
  const int lane = threadIdx.x % WAVE_SIZE;
  const int l16 = lane & 15;
  const int g4 = lane >> 4;
  const int d8 = D / 8;

  stage_tile<kSwizzle>(q_lds, q + (long)b * q_sb + (long)h * q_sh, q_sn,
                       qbase, kQB, Nq, D, d8);
  stage_tile<kSwizzle>(do_lds, dov + ((long)bh * Nq) * D, D, qbase, kQB, Nq, D, d8);

  // per-row softmax stats (4 rows per lane-group, same value in all 16 lanes)
  float lse_r[4], delta_r[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int qrow = qbase + wave * 16 + g4 * 4 + r;
    lse_r[r] = (qrow < Nq) ? lse[(long)bh * Nq + qrow] : INFINITY;
    delta_r[r] = (qrow < Nq) ? delta[(long)bh * Nq + qrow] : 0.f;
  }

  f32x4 acc_dq[kMaxD / 16];
#pragma unroll
  for (int f = 0; f < kMaxD / 16; ++f) acc_dq[f] = f32x4{0.f, 0.f, 0.f, 0.f};

  const __bf16* k_bh = k + (long)b * k_sb + (long)h * k_sh;
  const __bf16* v_bh = v + (long)b * v_sb + (long)h * v_sh;
  const int n_kv = (Nk + kKvB - 1) / kKvB;
  for (int kt = 0; kt < n_kv; ++kt) {
    const int kv0 = kt * kKvB;
    __syncthreads();  // previous iteration's reads done before restage
    stage_tile<kSwizzle>(k_lds, k_bh, k_sn, kv0, kKvB, Nk, D, d8);
    stage_tile<kSwizzle>(v_lds, v_bh, v_sn, kv0, kKvB, Nk, D, d8);
    if (kUseT) {
      stage_tile_t<kSwizzle>(kt_lds, k_bh, k_sn, kv0, kKvB, Nk, D, d8);
    }
    __syncthreads();

    // S and dP fragments: [n16][reg], rows = q (wave*16 + g4*4 + r), col = kv
    float ds_frag[2][4];
#pragma unroll
    for (int n16 = 0; n16 < 2; ++n16) {
      f32x4 acc_s = {0.f, 0.f, 0.f, 0.f};
      f32x4 acc_dp = {0.f, 0.f, 0.f, 0.f};
      for (int d0 = 0; d0 < D; d0 += 32) {
        bf16x8_t qa = frag_row<kSwizzle>(q_lds, wave * 16 + l16, d0, g4, D);
        bf16x8_t kb = frag_row<kSwizzle>(k_lds, n16 * 16 + l16, d0, g4, D);
        acc_s = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qa, kb, acc_s, 0, 0, 0);
        bf16x8_t da = frag_row<kSwizzle>(do_lds, wave * 16 + l16, d0, g4, D);
        bf16x8_t vb = frag_row<kSwizzle>(v_lds, n16 * 16 + l16, d0, g4, D);
        acc_dp = __builtin_amdgcn_mfma_f32_16x16x32_bf16(da, vb, acc_dp, 0, 0, 0);
      }
      const int kvcol = kv0 + n16 * 16 + l16;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qrow = qbase + wave * 16 + g4 * 4 + r;
        float s = acc_s[r] * scale;
        if (kvcol >= Nk || qrow >= Nq) {
          s = kNegInf;
        } else if (kHasMask) {
          s += mask[(long)(b % mB) * m_sb + (long)h * m_sh + (long)qrow * Nk + kvcol];
          if (s < kNegInf) s = kNegInf;
        }
        const float p = (s <= kNegInf) ? 0.f : __expf(s - lse_r[r]);
        ds_frag[n16][r] = p * (acc_dp[r] - delta_r[r]) * scale;
      }
    }

    // round-trip dS through LDS to re-shape C-layout -> A-fragment layout
#pragma unroll
    for (int n16 = 0; n16 < 2; ++n16) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        ds_lds[wave * (16 * kKvB) + (g4 * 4 + r) * kKvB + n16 * 16 + l16] =
            (__bf16)ds_frag[n16][r];
      }
    }
    __asm__ volatile("s_waitcnt lgkmcnt(0)" ::: "memory");

    // dQ += dS K : A = dS row fragment (k = kv, contiguous), B = K columns
    bf16x8_t dsa = *reinterpret_cast<const bf16x8_t*>(
        ds_lds + wave * (16 * kKvB) + l16 * kKvB + g4 * 8);
#pragma unroll
    for (int f = 0; f < kMaxD / 16; ++f) {
      if (f * 16 >= D) break;
      bf16x8_t kcol = kUseT
          ? frag_row_t<kSwizzle>(kt_lds, f * 16 + l16, g4, kKvB)
          : frag_col<kSwizzle>(k_lds, g4, f * 16 + l16, D);
      acc_dq[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dsa, kcol, acc_dq[f], 0, 0, 0);
    }
  }

  // write dQ (C layout: row = wave*16 + g4*4 + r, col = f*16 + l16)
  __bf16* dq_bh = dq + (long)b * dq_sb + (long)h * dq_sh;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int qrow = qbase + wave * 16 + g4 * 4 + r;
    if (qrow >= Nq) continue;
    for (int f = 0; f < kMaxD / 16; ++f) {
      if (f * 16 >= D) break;
      dq_bh[(long)qrow * dq_sn + f * 16 + l16] = (__bf16)acc_dq[f][r];
    }
  }
}

// ---------------------------------------------------------------------------
// dK/dV pass: block owns 64 kv rows, loops 32-row q tiles (transposed math).
// ---------------------------------------------------------------------------
template <int kMaxD, bool kHasMask, bool kSwizzle>
__global__ __launch_bounds__(kBlockThreads)
void attn_bwd_dkdv_kernel(
    const __bf16* __restrict__ q,
    const __bf16* __restrict__ k,
    const __bf16* __restrict__ v,
    const __bf16* __restrict__ dov,   // contiguous [B,H,Nq,D]
    const float* __restrict__ lse,
    const float* __restrict__ delta,
    const float* __restrict__ mask,
    __bf16* __restrict__ dk,
    __bf16* __restrict__ dv,
    int B, int H, int Nq, int Nk, int D, float scale,
    int mB, long m_sb, long m_sh,
    long q_sb, long q_sh, long q_sn,
    long k_sb, long k_sh, long k_sn,
    long v_sb, long v_sh, long v_sn,
    long dk_sb, long dk_sh, long dk_sn,
    long dv_sb, long dv_sh, long dv_sn) {
  constexpr int kKvB = 64, kQB = 32;
  constexpr bool kUseT = kMaxD <= 64;  // transposed Q/dO tiles fit in LDS
  __shared__ __bf16 k_lds[kKvB * kMaxD];
  __shared__ __bf16 v_lds[kKvB * kMaxD];
  __shared__ __bf16 q_lds[kQB * kMaxD];
  __shared__ __bf16 do_lds[kQB * kMaxD];
  __shared__ __bf16 pt_lds[4 * 16 * kQB];   // reused for P^T then dS^T
  __shared__ __bf16 qt_lds[kUseT ? kMaxD * kQB : 1];
  __shared__ __bf16 dot_lds[kUseT ? kMaxD * kQB : 1];

  const int bh = blockIdx.y;
  const int b = bh / H;
  const int h = bh % H;
  const int kvbase = blockIdx.x * kKvB;
  const int wave = threadIdx.x / WAVE_SIZE;
  const int lane = threadIdx.x % WAVE_SIZE;
  const int l16 = lane & 15;
  const int g4 = lane >> 4;
  const int d8 = D / 8;

  stage_tile<kSwizzle>(k_lds, k + (long)b * k_sb + (long)h * k_sh, k_sn,
                       kvbase, kKvB, Nk, D, d8);
  stage_tile<kSwizzle>(v_lds, v + (long)b * v_sb + (long)h * v_sh, v_sn,
                       kvbase, kKvB, Nk, D, d8);

  f32x4 acc_dk[kMaxD / 16], acc_dv[kMaxD / 16];
#pragma unroll
  for (int f = 0; f < kMaxD / 16; ++f) {
    acc_dk[f] = f32x4{0.f, 0.f, 0.f, 0.f};
    acc_dv[f] = f32x4{0.f, 0.f, 0.f, 0.f};
  }

  const __bf16* q_bh = q + (long)b * q_sb + (long)h * q_sh;
  const __bf16* do_bh = dov + ((long)bh * Nq) * D;
  const int n_qt = (Nq + kQB - 1) / kQB;
  for (int qt = 0; qt < n_qt; ++qt) {
    const int q0 = qt * kQB;
    __syncthreads();
    stage_tile<kSwizzle>(q_lds, q_bh, q_sn, q0, kQB, Nq, D, d8);
    stage_tile<kSwizzle>(do_lds, do_bh, D, q0, kQB, Nq, D, d8);
    if (kUseT) {
      stage_tile_t<kSwizzle>(qt_lds, q_bh, q_sn, q0, kQB, Nq, D, d8);
      stage_tile_t<kSwizzle>(dot_lds, do_bh, D, q0, kQB, Nq, D, d8);
    }
    __syncthreads();

    // per-q-column stats: col = q0 + n16*16 + l16
    // S^T = K Q^T, dP^T = V dO^T : rows = kv (wave*16 + g4*4 + r), col = q
    float pt_frag[2][4], dst_frag[2][4];
#pragma unroll
    for (int n16 = 0; n16 < 2; ++n16) {
      const int qcol = q0 + n16 * 16 + l16;
      const float lse_c = (qcol < Nq) ? lse[(long)bh * Nq + qcol] : INFINITY;
      const float delta_c = (qcol < Nq) ? delta[(long)bh * Nq + qcol] : 0.f;
      f32x4 acc_s = {0.f, 0.f, 0.f, 0.f};
      f32x4 acc_dp = {0.f, 0.f, 0.f, 0.f};
      for (int d0 = 0; d0 < D; d0 += 32) {
        bf16x8_t ka = frag_row<kSwizzle>(k_lds, wave * 16 + l16, d0, g4, D);
        bf16x8_t qb = frag_row<kSwizzle>(q_lds, n16 * 16 + l16, d0, g4, D);
        acc_s = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ka, qb, acc_s, 0, 0, 0);
        bf16x8_t va = frag_row<kSwizzle>(v_lds, wave * 16 + l16, d0, g4, D);
        bf16x8_t db = frag_row<kSwizzle>(do_lds, n16 * 16 + l16, d0, g4, D);
        acc_dp = __builtin_amdgcn_mfma_f32_16x16x32_bf16(va, db, acc_dp, 0, 0, 0);
      }
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int kvrow = kvbase + wave * 16 + g4 * 4 + r;
        float s = acc_s[r] * scale;
        if (kvrow >= Nk || qcol >= Nq) {
          s = kNegInf;
        } else if (kHasMask) {
          s += mask[(long)(b % mB) * m_sb + (long)h * m_sh + (long)qcol * Nk + kvrow];
          if (s < kNegInf) s = kNegInf;
        }
        const float p = (s <= kNegInf) ? 0.f : __expf(s - lse_c);
        pt_frag[n16][r] = p;
        dst_frag[n16][r] = p * (acc_dp[r] - delta_c) * scale;
      }
    }

    // dV += P^T dO : P^T through LDS, dO columns gathered
#pragma unroll
    for (int n16 = 0; n16 < 2; ++n16) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        pt_lds[wave * (16 * kQB) + (g4 * 4 + r) * kQB + n16 * 16 + l16] =
            (__bf16)pt_frag[n16][r];
      }
    }
    __asm__ volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    {
      bf16x8_t pa = *reinterpret_cast<const bf16x8_t*>(
          pt_lds + wave * (16 * kQB) + l16 * kQB + g4 * 8);
#pragma unroll
      for (int f = 0; f < kMaxD / 16; ++f) {
        if (f * 16 >= D) break;
        bf16x8_t dcol = kUseT
            ? frag_row_t<kSwizzle>(dot_lds, f * 16 + l16, g4, kQB)
            : frag_col<kSwizzle>(do_lds, g4, f * 16 + l16, D);
        acc_dv[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pa, dcol, acc_dv[f], 0, 0, 0);
      }
    }
    __syncthreads();  // pt_lds reuse

    // dK += dS^T Q
#pragma unroll
    for (int n16 = 0; n16 < 2; ++n16) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        pt_lds[wave * (16 * kQB) + (g4 * 4 + r) * kQB + n16 * 16 + l16] =
            (__bf16)dst_frag[n16][r];
      }
    }
    __asm__ volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    {
      bf16x8_t dsa = *reinterpret_cast<const bf16x8_t*>(
          pt_lds + wave * (16 * kQB) + l16 * kQB + g4 * 8);
#pragma unroll
      for (int f = 0; f < kMaxD / 16; ++f) {
        if (f * 16 >= D) break;
        bf16x8_t qcolf = kUseT
            ? frag_row_t<kSwizzle>(qt_lds, f * 16 + l16, g4, kQB)
            : frag_col<kSwizzle>(q_lds, g4, f * 16 + l16, D);
        acc_dk[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dsa, qcolf, acc_dk[f], 0, 0, 0);
      }
    }
  }

  // write dK, dV (rows = kv)
  __bf16* dk_bh = dk + (long)b * dk_sb + (long)h * dk_sh;
  __bf16* dv_bh = dv + (long)b * dv_sb + (long)h * dv_sh;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int kvrow = kvbase + wave * 16 + g4 * 4 + r;
    if (kvrow >= Nk) continue;
    for (int f = 0; f < kMaxD / 16; ++f) {
      if (f * 16 >= D) break;
      dk_bh[(long)kvrow * dk_sn + f * 16 + l16] = (__bf16)acc_dk[f][r];
      dv_bh[(long)kvrow * dv_sn + f * 16 + l16] = (__bf16)acc_dv[f][r];
    }
  }
}

struct MaskInfo {
  const float* ptr = nullptr;
  int mB = 1;
  long sb = 0, sh = 0;
};

MaskInfo mask_info(const c10::optional<at::Tensor>& mask, int Nq, int Nk) {
  MaskInfo mi;
  if (mask.has_value()) {
    mi.ptr = mask->data_ptr<float>();
    mi.mB = (int)mask->size(0);
    mi.sb = mi.mB == 1 ? 0 : mask->size(1) * (long)Nq * Nk;
    mi.sh = mask->size(1) == 1 ? 0 : (long)Nq * Nk;
  }
  return mi;
}

template <int kMaxD>
void launch_attn_bwd(
    const at::Tensor& q, const at::Tensor& k, const at::Tensor& v,
    const at::Tensor& dov, const at::Tensor& lse, const at::Tensor& delta,
    const c10::optional<at::Tensor>& mask,
    at::Tensor& dq, at::Tensor& dk, at::Tensor& dv,
    int B, int H, int Nq, int Nk, int D, float scale, hipStream_t stream) {
  const MaskInfo mi = mask_info(mask, Nq, Nk);
  const bool swizzle = (D == 64 || D == 128);
  dim3 block(kBlockThreads);
  dim3 grid_dq(cdiv(Nq, 64), B * H);
  dim3 grid_kv(cdiv(Nk, 64), B * H);

  auto run_dq = [&](auto kern) {
    hipLaunchKernelGGL(kern, grid_dq, block, 0, stream,
        (const __bf16*)q.data_ptr(), (const __bf16*)k.data_ptr(), (const __bf16*)v.data_ptr(),
        (const __bf16*)dov.data_ptr(), lse.data_ptr<float>(), delta.data_ptr<float>(),
        mi.ptr, (__bf16*)dq.data_ptr(),
        B, H, Nq, Nk, D, scale, mi.mB, mi.sb, mi.sh,
        q.stride(0), q.stride(1), q.stride(2),
        k.stride(0), k.stride(1), k.stride(2),
        v.stride(0), v.stride(1), v.stride(2),
        dq.stride(0), dq.stride(1), dq.stride(2));
  };
  auto run_kv = [&](auto kern) {
    hipLaunchKernelGGL(kern, grid_kv, block, 0, stream,
        (const __bf16*)q.data_ptr(), (const __bf16*)k.data_ptr(), (const __bf16*)v.data_ptr(),
        (const __bf16*)dov.data_ptr(), lse.data_ptr<float>(), delta.data_ptr<float>(),
        mi.ptr, (__bf16*)dk.data_ptr(), (__bf16*)dv.data_ptr(),
        B, H, Nq, Nk, D, scale, mi.mB, mi.sb, mi.sh,
        q.stride(0), q.stride(1), q.stride(2),
        k.stride(0), k.stride(1), k.stride(2),
        v.stride(0), v.stride(1), v.stride(2),
        dk.stride(0), dk.stride(1), dk.stride(2),
        dv.stride(0), dv.stride(1), dv.stride(2));
  };
  const bool has_mask = mask.has_value();
  if (has_mask && swizzle) {
    run_dq(attn_bwd_dq_kernel<kMaxD, true, true>);
    run_kv(attn_bwd_dkdv_kernel<kMaxD, true, true>);
  } else if (has_mask) {
    run_dq(attn_bwd_dq_kernel<kMaxD, true, false>);
    run_kv(attn_bwd_dkdv_kernel<kMaxD, true, false>);
  } else if (swizzle) {
    run_dq(attn_bwd_dq_kernel<kMaxD, false, true>);
    run_kv(attn_bwd_dkdv_kernel<kMaxD, false, true>);
  } else {
    run_dq(attn_bwd_dq_kernel<kMaxD, false, false>);
    run_kv(attn_bwd_dkdv_kernel<kMaxD, false, false>);
  }
  HIP_CHECK_LAST();
}

}  // namespace

// Full fused backward.  q/k/v strided [B,H,N,D] (head_dim contiguous);
// dov contiguous [B,H,Nq,D]; lse/delta fp32 [B,H,Nq].
// dq/dk/dv are preallocated by the caller (possibly views into a packed
// [B,N,3,H,D] dqkv buffer) with head_dim contiguous.
void attention_bwd(at::Tensor q, at::Tensor k, at::Tensor v,
                   at::Tensor dov, at::Tensor lse, at::Tensor delta,
                   c10::optional<at::Tensor> mask,
                   at::Tensor dq, at::Tensor dk, at::Tensor dv,
                   double scale) {
  TORCH_CHECK(q.is_cuda() && q.dim() == 4 && q.scalar_type() == at::kBFloat16);
  TORCH_CHECK(q.stride(3) == 1 && k.stride(3) == 1 && v.stride(3) == 1 &&
              dq.stride(3) == 1 && dk.stride(3) == 1 && dv.stride(3) == 1,
              "attention_bwd: head_dim stride must be 1");
  TORCH_CHECK(dov.is_contiguous(), "attention_bwd: dO must be contiguous");
  int B = q.size(0), H = q.size(1), Nq = q.size(2), D = q.size(3);
  int Nk = k.size(2);
  TORCH_CHECK(D % 32 == 0 && D <= 128, "attention_bwd: head_dim multiple of 32, <=128");
  if (mask.has_value()) {
    TORCH_CHECK(mask->is_contiguous() && mask->scalar_type() == at::kFloat);
    TORCH_CHECK(B % mask->size(0) == 0 && (mask->size(1) == H || mask->size(1) == 1) &&
                mask->size(2) == Nq && mask->size(3) == Nk);
  }
  auto stream = at::hip::getCurrentHIPStream();
  if (D <= 32) {
    launch_attn_bwd<32>(q, k, v, dov, lse, delta, mask, dq, dk, dv, B, H, Nq, Nk, D,
                        (float)scale, stream);
  } else if (D <= 64) {
    launch_attn_bwd<64>(q, k, v, dov, lse, delta, mask, dq, dk, dv, B, H, Nq, Nk, D,
                        (float)scale, stream);
  } else if (D <= 96) {
    launch_attn_bwd<96>(q, k, v, dov, lse, delta, mask, dq, dk, dv, B, H, Nq, Nk, D,
                        (float)scale, stream);
  } else {
    launch_attn_bwd<128>(q, k, v, dov, lse, delta, mask, dq, dk, dv, B, H, Nq, Nk, D,
                         (float)scale, stream);
  }
}
