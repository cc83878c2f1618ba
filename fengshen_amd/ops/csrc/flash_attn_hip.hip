#include "hip/hip_runtime.h"
// Flash-style fused causal attention FORWARD for MI355X (gfx950).
//
// Replaces the reference's flash_attn_cuda import (flash_attention.py:7) with
// a native CDNA4 kernel: per Q-tile, stream K/V tiles through LDS, QK^T and
// PV on MFMA (v_mfma_f32_16x16x32_bf16), online softmax in fp32 registers.
// No S×S materialization (removes the reference's fused-softmax sk<=2048 cap
// and the 3x HBM round trip of the bmm path).
//
// v2 structure:
//   block = 4 waves; wave w owns 16 q-rows  => QBLK = 64
//   KVBLK = 64; K staged [64][D] with XOR swizzle (guide §6 G4: row-major
//   [*][128] bf16 is a 16-way ds_read_b128 bank conflict; byte ^= (row&7)<<4
//   fixes it); V staged transposed [D][64+pad] so PV B-fragments read
//   16B-contiguous kv runs.  D = 128, bf16, causal, seq % 64 == 0.
// Outputs: O [b,h,s,D] and LSE [b,h,s] (softmax log-sum-exp, for backward).

#include "common.h"

typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define FA_D 128
#define FA_QBLK 128          // 8 waves x 16 q rows: 2x compute per barrier
#define FA_KVBLK 64
#define FA_WAVES 8
#define FA_VPAD 8

// XOR swizzle for k_lds rows (applies to 16B-aligned byte offsets)
__device__ __forceinline__ int kswz(int row, int byte_in_row) {
  return byte_in_row ^ ((row & 7) << 4);
}

__global__ __launch_bounds__(FA_WAVES * 64)
void flash_attn_fwd_kernel(const bf16_t* __restrict__ Q,
                           const bf16_t* __restrict__ K,
                           const bf16_t* __restrict__ V,
                           bf16_t* __restrict__ O,
                           float* __restrict__ LSE,
                           int b, int h, int s, float scale) {
  // raw bf16 bits in short storage (short=bf16_t assignment would convert)
  __shared__ short k_lds[FA_KVBLK][FA_D];              // swizzled rows
  __shared__ short vt_lds[FA_D][FA_KVBLK + FA_VPAD];   // V transposed
  __shared__ short p_lds[FA_WAVES][16][FA_KVBLK + FA_VPAD];

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int qb = blockIdx.x;
  const int head = blockIdx.y;
  const int batch = blockIdx.z;

  const long bh_off = ((long)batch * h + head) * s * FA_D;
  const bf16_t* Qp = Q + bh_off;
  const bf16_t* Kp = K + bh_off;
  const bf16_t* Vp = V + bh_off;
  bf16_t* Op = O + bh_off;

  const int q0 = qb * FA_QBLK + wave * 16;
  // partial blocks (s % FA_QBLK != 0): OOB waves compute on clamped rows
  // (must still hit every __syncthreads) and skip their stores
  const bool q_active = q0 < s;
  const int q0c = q_active ? q0 : s - 16;

  // ---- Q tile -> A-fragments (layout LA0: row=l%16, k=(l/16)*8+j;
  // verified on HW by scripts/mfma_probe), pre-scaled -----------------------
  bf16x8 q_frag[4];
  {
    const int row = lane & 15;
    const int k0 = (lane >> 4) * 8;
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      bf16x8 raw = *reinterpret_cast<const bf16x8*>(
          Qp + (long)(q0c + row) * FA_D + c * 32 + k0);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        unsigned short u = (unsigned short)raw[j];
        float f = __uint_as_float(((unsigned int)u) << 16) * scale;
        raw[j] = (short)__hip_bfloat16_raw(__float2bfloat16(f)).x;
      }
      q_frag[c] = raw;
    }
  }

  float m_run[4], l_run[4];
  f32x4 o_acc[8];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    m_run[r] = -INFINITY;
    l_run[r] = 0.f;
  }
#pragma unroll
  for (int t = 0; t < 8; ++t) o_acc[t] = f32x4{0, 0, 0, 0};

  const int q_hi = min(qb * FA_QBLK + FA_QBLK, s) - 1;
  const int n_kv_tiles = (q_hi / FA_KVBLK) + 1;

  // async-STAGE split (guide §6 G15): global loads for tile t+1 are issued
  // DURING tile t's compute (registers k_reg/v_reg), and only the cheap
  // LDS writes sit between the barriers — HBM latency hides under MFMA.
  const int tid = threadIdx.x;
  bf16x8 k_reg[2], v_reg[2];
#pragma unroll
  for (int sweep = 0; sweep < 2; ++sweep) {
    const int i = tid * 8 + sweep * 4096;
    const int kr = i / FA_D;
    const int kc = i % FA_D;
    k_reg[sweep] = *reinterpret_cast<const bf16x8*>(Kp + (long)kr * FA_D + kc);
    v_reg[sweep] = *reinterpret_cast<const bf16x8*>(Vp + (long)kr * FA_D + kc);
  }

  for (int kt = 0; kt < n_kv_tiles; ++kt) {
    const int k_base = kt * FA_KVBLK;
    __syncthreads();
#pragma unroll
    for (int sweep = 0; sweep < 2; ++sweep) {
      const int i = tid * 8 + sweep * 4096;
      const int kr = i / FA_D;
      const int kc = i % FA_D;
      *reinterpret_cast<bf16x8*>(
          reinterpret_cast<char*>(&k_lds[kr][0]) + kswz(kr, kc * 2)) =
          k_reg[sweep];
      bf16x8 vv = v_reg[sweep];
      // kv index XOR-swizzled by d bits 3-5 (the transpose scatter would
      // otherwise put 16 lanes of one K-row into ONE bank)
#pragma unroll
      for (int j = 0; j < 8; ++j)
        vt_lds[kc + j][kr ^ ((kc + j) & 0x38)] = vv[j];
    }
    __syncthreads();
    if (kt + 1 < n_kv_tiles) {
      const int nb = (kt + 1) * FA_KVBLK;
#pragma unroll
      for (int sweep = 0; sweep < 2; ++sweep) {
        const int i = tid * 8 + sweep * 4096;
        const int kr = i / FA_D;
        const int kc = i % FA_D;
        k_reg[sweep] = *reinterpret_cast<const bf16x8*>(
            Kp + (long)(nb + kr) * FA_D + kc);
        v_reg[sweep] = *reinterpret_cast<const bf16x8*>(
            Vp + (long)(nb + kr) * FA_D + kc);
      }
    }

    // ---- S = (sQ) @ K^T : four 16x16 n-tiles ------------------------------
    f32x4 s_acc[4];
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) s_acc[nt] = f32x4{0, 0, 0, 0};
    {
      const int col = lane & 15;
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
        const int krow = nt * 16 + col;
#pragma unroll
        for (int c = 0; c < 4; ++c) {
          const int d0 = c * 32 + (lane >> 4) * 8;
          bf16x8 bfrag = *reinterpret_cast<const bf16x8*>(
              reinterpret_cast<char*>(&k_lds[krow][0]) + kswz(krow, d0 * 2));
          s_acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              q_frag[c], bfrag, s_acc[nt], 0, 0, 0);
        }
      }
    }

    // ---- causal mask + online softmax -------------------------------------
    const int col = lane & 15;
    float p[4][4];
    float tile_max[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int qrow = q0c + (lane >> 4) * 4 + r;
      float tm = -INFINITY;
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
        const int kcol = k_base + nt * 16 + col;
        float v = s_acc[nt][r];
        if (kcol > qrow) v = -INFINITY;
        p[nt][r] = v;
        tm = fmaxf(tm, v);
      }
      tile_max[r] = tm;
    }
#pragma unroll
    for (int off = 8; off > 0; off >>= 1) {
#pragma unroll
      for (int r = 0; r < 4; ++r)
        tile_max[r] = fmaxf(tile_max[r], __shfl_xor(tile_max[r], off, 64));
    }
    float alpha[4], rowsum[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const float m_new = fmaxf(m_run[r], tile_max[r]);
      alpha[r] = (m_run[r] == -INFINITY) ? 0.f : __expf(m_run[r] - m_new);
      m_run[r] = m_new;
      float ps = 0.f;
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
        float e = (p[nt][r] == -INFINITY) ? 0.f : __expf(p[nt][r] - m_new);
        p[nt][r] = e;
        ps += e;
      }
      rowsum[r] = ps;
    }
#pragma unroll
    for (int off = 8; off > 0; off >>= 1) {
#pragma unroll
      for (int r = 0; r < 4; ++r)
        rowsum[r] += __shfl_xor(rowsum[r], off, 64);
    }
#pragma unroll
    for (int r = 0; r < 4; ++r)
      l_run[r] = l_run[r] * alpha[r] + rowsum[r];
#pragma unroll
    for (int t = 0; t < 8; ++t) {
#pragma unroll
      for (int r = 0; r < 4; ++r) o_acc[t][r] *= alpha[r];
    }

    // ---- P -> per-wave LDS, then PV ---------------------------------------
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = (lane >> 4) * 4 + r;
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
        p_lds[wave][row][nt * 16 + col] =
            (short)__hip_bfloat16_raw(__float2bfloat16(p[nt][r])).x;
      }
    }
    __builtin_amdgcn_s_waitcnt(0);  // wave-local LDS visibility

    // A-frags over kv (two K=32 chunks), B-frags from V^T
#pragma unroll
    for (int kc = 0; kc < 2; ++kc) {
      const int row = lane & 15;
      const int kv0 = kc * 32 + (lane >> 4) * 8;
      bf16x8 pa = *reinterpret_cast<const bf16x8*>(&p_lds[wave][row][kv0]);
#pragma unroll
      for (int t = 0; t < 8; ++t) {
        const int dcol = t * 16 + (lane & 15);
        bf16x8 bfrag = *reinterpret_cast<const bf16x8*>(
            &vt_lds[dcol][kv0 ^ (dcol & 0x38)]);
        o_acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pa, bfrag, o_acc[t],
                                                           0, 0, 0);
      }
    }
  }

  // ---- epilogue -----------------------------------------------------------
  if (!q_active) return;  // after the last barrier: safe to exit
  const int col = lane & 15;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int qrow = q0 + (lane >> 4) * 4 + r;
    const float inv_l = (l_run[r] > 0.f) ? 1.f / l_run[r] : 0.f;
#pragma unroll
    for (int t = 0; t < 8; ++t) {
      Op[(long)qrow * FA_D + t * 16 + col] =
          __float2bfloat16(o_acc[t][r] * inv_l);
    }
    if (col == 0 && LSE) {
      LSE[((long)batch * h + head) * s + qrow] =
          m_run[r] + logf(fmaxf(l_run[r], 1e-30f));
    }
  }
}

extern "C" void fs_flash_attn_fwd(const void* q, const void* k, const void* v,
                                  void* o, float* lse, int b, int h, int s,
                                  float scale, hipStream_t stream) {
  dim3 grid((s + FA_QBLK - 1) / FA_QBLK, h, b);
  dim3 block(FA_WAVES * 64);
  hipLaunchKernelGGL(flash_attn_fwd_kernel, grid, block, 0, stream,
                     (const bf16_t*)q, (const bf16_t*)k, (const bf16_t*)v,
                     (bf16_t*)o, lse, b, h, s, scale);
}

// ===========================================================================
// BACKWARD
// ===========================================================================
// FlashAttention-2-style: delta = rowsum(dO*O); then
//   dQ kernel (q-major):   S=(sQ)K^T; P=exp(S-LSE); dP=dO V^T;
//                          dS=P*(dP-delta)*s; dQ += dS K
//   dKdV kernel (kv-major): St=(sK)Q^T; Pt=exp(St-LSE[col]); dPt=V dO^T;
//                          dV += Pt dO; dSt=Pt*(dPt-delta[col])*s; dK += dSt Q
// Same fragment layouts / swizzles as forward (HW-verified via mfma_probe).
// Streaming tiles are 32-wide (FB_T) so LDS stays ~40KB -> 3-4 blocks/CU
// (the 64-wide first cut sat at 1 block/CU and was latency-bound).

#define FB_T 32

__global__ __launch_bounds__(256)
void flash_delta_kernel(const bf16_t* __restrict__ dO,
                        const bf16_t* __restrict__ O,
                        float* __restrict__ delta, long rows) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const long row0 = (long)blockIdx.x * 4 + wid;
  const long stride = (long)gridDim.x * 4;
  for (long row = row0; row < rows; row += stride) {
    const bf16_t* dop = dO + row * FA_D + lane * 2;
    const bf16_t* op = O + row * FA_D + lane * 2;
    float sacc = 0.f;
#pragma unroll
    for (int j = 0; j < 2; ++j)
      sacc += __bfloat162float(dop[j]) * __bfloat162float(op[j]);
    sacc = wave_reduce_sum(sacc);
    if (lane == 0) delta[row] = sacc;
  }
}

// Q-major: each block owns 64 q rows (wave -> 16), streams KV in 32-tiles.
__global__ __launch_bounds__(FA_WAVES * 64)
void flash_attn_bwd_dq_kernel(const bf16_t* __restrict__ Q,
                              const bf16_t* __restrict__ K,
                              const bf16_t* __restrict__ V,
                              const bf16_t* __restrict__ dO,
                              const float* __restrict__ LSE,
                              const float* __restrict__ Delta,
                              bf16_t* __restrict__ dQ,
                              int b, int h, int s, float scale) {
  __shared__ short k_lds[FB_T][FA_D];                 // K rows, swizzled
  __shared__ short v_lds[FB_T][FA_D];                 // V rows, swizzled
  __shared__ short kt_lds[FA_D][FB_T + FA_VPAD];      // K^T, kv-swizzled
  __shared__ short p_lds[FA_WAVES][16][FB_T + FA_VPAD];

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const long bh = (long)blockIdx.z * h + blockIdx.y;
  const bf16_t* Qp = Q + bh * s * FA_D;
  const bf16_t* Kp = K + bh * s * FA_D;
  const bf16_t* Vp = V + bh * s * FA_D;
  const bf16_t* dOp = dO + bh * s * FA_D;
  bf16_t* dQp = dQ + bh * s * FA_D;
  const float* lse = LSE + bh * s;
  const float* dlt = Delta + bh * s;

  const int q0 = blockIdx.x * FA_QBLK + wave * 16;
  const bool q_active = q0 < s;
  const int q0c = q_active ? q0 : s - 16;

  bf16x8 q_frag[4], do_frag[4];
  {
    const int row = lane & 15;
    const int k0 = (lane >> 4) * 8;
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      bf16x8 raw = *reinterpret_cast<const bf16x8*>(
          Qp + (long)(q0c + row) * FA_D + c * 32 + k0);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        unsigned short u = (unsigned short)raw[j];
        float f = __uint_as_float(((unsigned int)u) << 16) * scale;
        raw[j] = (short)__hip_bfloat16_raw(__float2bfloat16(f)).x;
      }
      q_frag[c] = raw;
      do_frag[c] = *reinterpret_cast<const bf16x8*>(
          dOp + (long)(q0c + row) * FA_D + c * 32 + k0);
    }
  }
  float lse_r[4], dlt_r[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int qrow = q0c + (lane >> 4) * 4 + r;
    lse_r[r] = lse[qrow];
    dlt_r[r] = dlt[qrow];
  }

  f32x4 dq_acc[8];
#pragma unroll
  for (int t = 0; t < 8; ++t) dq_acc[t] = f32x4{0, 0, 0, 0};

  const int q_hi = min(blockIdx.x * FA_QBLK + FA_QBLK, s) - 1;
  const int n_kv_tiles = (q_hi / FB_T) + 1;

  for (int kt = 0; kt < n_kv_tiles; ++kt) {
    const int k_base = kt * FB_T;
    __syncthreads();
    {
      const int tid = threadIdx.x;
      for (int i = tid * 8; i < FB_T * FA_D; i += FA_WAVES * 64 * 8) {
        const int kr = i / FA_D;
        const int kc = i % FA_D;
        bf16x8 kk = *reinterpret_cast<const bf16x8*>(
            Kp + (long)(k_base + kr) * FA_D + kc);
        *reinterpret_cast<bf16x8*>(
            reinterpret_cast<char*>(&k_lds[kr][0]) + kswz(kr, kc * 2)) = kk;
        bf16x8 vv = *reinterpret_cast<const bf16x8*>(
            Vp + (long)(k_base + kr) * FA_D + kc);
        *reinterpret_cast<bf16x8*>(
            reinterpret_cast<char*>(&v_lds[kr][0]) + kswz(kr, kc * 2)) = vv;
#pragma unroll
        for (int j = 0; j < 8; ++j)
          kt_lds[kc + j][kr ^ ((kc + j) & 0x18)] = kk[j];
      }
    }
    __syncthreads();

    f32x4 s_acc[2], dp_acc[2];
#pragma unroll
    for (int nt = 0; nt < 2; ++nt) {
      s_acc[nt] = f32x4{0, 0, 0, 0};
      dp_acc[nt] = f32x4{0, 0, 0, 0};
    }
    {
      const int col = lane & 15;
#pragma unroll
      for (int nt = 0; nt < 2; ++nt) {
        const int krow = nt * 16 + col;
#pragma unroll
        for (int c = 0; c < 4; ++c) {
          const int d0 = c * 32 + (lane >> 4) * 8;
          bf16x8 kb = *reinterpret_cast<const bf16x8*>(
              reinterpret_cast<char*>(&k_lds[krow][0]) + kswz(krow, d0 * 2));
          s_acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              q_frag[c], kb, s_acc[nt], 0, 0, 0);
          bf16x8 vb = *reinterpret_cast<const bf16x8*>(
              reinterpret_cast<char*>(&v_lds[krow][0]) + kswz(krow, d0 * 2));
          dp_acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              do_frag[c], vb, dp_acc[nt], 0, 0, 0);
        }
      }
    }

    const int col = lane & 15;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int qrow = q0c + (lane >> 4) * 4 + r;
      const int row = (lane >> 4) * 4 + r;
#pragma unroll
      for (int nt = 0; nt < 2; ++nt) {
        const int kcol = k_base + nt * 16 + col;
        float pv = (kcol > qrow) ? 0.f : __expf(s_acc[nt][r] - lse_r[r]);
        float ds = pv * (dp_acc[nt][r] - dlt_r[r]) * scale;
        p_lds[wave][row][nt * 16 + col] =
            (short)__hip_bfloat16_raw(__float2bfloat16(ds)).x;
      }
    }
    __builtin_amdgcn_s_waitcnt(0);

    {
      const int row = lane & 15;
      const int kv0 = (lane >> 4) * 8;
      bf16x8 pa = *reinterpret_cast<const bf16x8*>(&p_lds[wave][row][kv0]);
#pragma unroll
      for (int t = 0; t < 8; ++t) {
        const int dcol = t * 16 + (lane & 15);
        bf16x8 bfrag = *reinterpret_cast<const bf16x8*>(
            &kt_lds[dcol][kv0 ^ (dcol & 0x18)]);
        dq_acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            pa, bfrag, dq_acc[t], 0, 0, 0);
      }
    }
  }

  if (!q_active) return;
  const int col = lane & 15;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int qrow = q0 + (lane >> 4) * 4 + r;
#pragma unroll
    for (int t = 0; t < 8; ++t)
      dQp[(long)qrow * FA_D + t * 16 + col] = __float2bfloat16(dq_acc[t][r]);
  }
}

// KV-major: each block owns 64 kv rows (wave -> 16), streams Q in 32-tiles.
__global__ __launch_bounds__(FA_WAVES * 64)
void flash_attn_bwd_dkv_kernel(const bf16_t* __restrict__ Q,
                               const bf16_t* __restrict__ K,
                               const bf16_t* __restrict__ V,
                               const bf16_t* __restrict__ dO,
                               const float* __restrict__ LSE,
                               const float* __restrict__ Delta,
                               bf16_t* __restrict__ dK,
                               bf16_t* __restrict__ dV,
                               int b, int h, int s, float scale) {
  __shared__ short q_lds[FB_T][FA_D];
  __shared__ short do_lds[FB_T][FA_D];
  __shared__ short qt_lds[FA_D][FB_T + FA_VPAD];
  __shared__ short dot_lds[FA_D][FB_T + FA_VPAD];
  __shared__ short p_lds[FA_WAVES][16][FB_T + FA_VPAD];

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const long bh = (long)blockIdx.z * h + blockIdx.y;
  const bf16_t* Qp = Q + bh * s * FA_D;
  const bf16_t* Kp = K + bh * s * FA_D;
  const bf16_t* Vp = V + bh * s * FA_D;
  const bf16_t* dOp = dO + bh * s * FA_D;
  bf16_t* dKp = dK + bh * s * FA_D;
  bf16_t* dVp = dV + bh * s * FA_D;
  const float* lse = LSE + bh * s;
  const float* dlt = Delta + bh * s;

  // each block owns FA_WAVES*16 kv rows (grid is s / FA_QBLK with
  // FA_QBLK == FA_WAVES*16)
  const int kv0_blk = blockIdx.x * (FA_WAVES * 16);
  const int kv0_wave = kv0_blk + wave * 16;
  const bool kv_active = kv0_wave < s;
  const int kv0c = kv_active ? kv0_wave : s - 16;

  bf16x8 k_frag[4], v_frag[4];
  {
    const int row = lane & 15;
    const int c0 = (lane >> 4) * 8;
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      bf16x8 raw = *reinterpret_cast<const bf16x8*>(
          Kp + (long)(kv0c + row) * FA_D + c * 32 + c0);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        unsigned short u = (unsigned short)raw[j];
        float f = __uint_as_float(((unsigned int)u) << 16) * scale;
        raw[j] = (short)__hip_bfloat16_raw(__float2bfloat16(f)).x;
      }
      k_frag[c] = raw;
      v_frag[c] = *reinterpret_cast<const bf16x8*>(
          Vp + (long)(kv0c + row) * FA_D + c * 32 + c0);
    }
  }

  f32x4 dv_acc[8], dk_acc[8];
#pragma unroll
  for (int t = 0; t < 8; ++t) {
    dv_acc[t] = f32x4{0, 0, 0, 0};
    dk_acc[t] = f32x4{0, 0, 0, 0};
  }

  const int first_qt = kv0_blk / FB_T;
  const int n_q_tiles = s / FB_T;

  for (int qt = first_qt; qt < n_q_tiles; ++qt) {
    const int q_base = qt * FB_T;
    __syncthreads();
    {
      const int tid = threadIdx.x;
      for (int i = tid * 8; i < FB_T * FA_D; i += FA_WAVES * 64 * 8) {
        const int qr = i / FA_D;
        const int qc = i % FA_D;
        bf16x8 qq = *reinterpret_cast<const bf16x8*>(
            Qp + (long)(q_base + qr) * FA_D + qc);
        *reinterpret_cast<bf16x8*>(
            reinterpret_cast<char*>(&q_lds[qr][0]) + kswz(qr, qc * 2)) = qq;
        bf16x8 dd = *reinterpret_cast<const bf16x8*>(
            dOp + (long)(q_base + qr) * FA_D + qc);
        *reinterpret_cast<bf16x8*>(
            reinterpret_cast<char*>(&do_lds[qr][0]) + kswz(qr, qc * 2)) = dd;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          qt_lds[qc + j][qr ^ ((qc + j) & 0x18)] = qq[j];
          dot_lds[qc + j][qr ^ ((qc + j) & 0x18)] = dd[j];
        }
      }
    }
    __syncthreads();

    f32x4 st_acc[2], dpt_acc[2];
#pragma unroll
    for (int nt = 0; nt < 2; ++nt) {
      st_acc[nt] = f32x4{0, 0, 0, 0};
      dpt_acc[nt] = f32x4{0, 0, 0, 0};
    }
    {
      const int col = lane & 15;
#pragma unroll
      for (int nt = 0; nt < 2; ++nt) {
        const int qrow = nt * 16 + col;
#pragma unroll
        for (int c = 0; c < 4; ++c) {
          const int d0 = c * 32 + (lane >> 4) * 8;
          bf16x8 qb = *reinterpret_cast<const bf16x8*>(
              reinterpret_cast<char*>(&q_lds[qrow][0]) + kswz(qrow, d0 * 2));
          st_acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              k_frag[c], qb, st_acc[nt], 0, 0, 0);
          bf16x8 db = *reinterpret_cast<const bf16x8*>(
              reinterpret_cast<char*>(&do_lds[qrow][0]) + kswz(qrow, d0 * 2));
          dpt_acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              v_frag[c], db, dpt_acc[nt], 0, 0, 0);
        }
      }
    }

    const int col = lane & 15;
    float pt[2][4];
#pragma unroll
    for (int nt = 0; nt < 2; ++nt) {
      const int qcol = q_base + nt * 16 + col;
      const float lse_c = lse[qcol];
      const float dlt_c = dlt[qcol];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int kvrow = kv0c + (lane >> 4) * 4 + r;
        float pv = (qcol < kvrow) ? 0.f : __expf(st_acc[nt][r] - lse_c);
        pt[nt][r] = pv;
        dpt_acc[nt][r] = pv * (dpt_acc[nt][r] - dlt_c) * scale;  // = dSt
      }
    }

#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = (lane >> 4) * 4 + r;
#pragma unroll
      for (int nt = 0; nt < 2; ++nt)
        p_lds[wave][row][nt * 16 + col] =
            (short)__hip_bfloat16_raw(__float2bfloat16(pt[nt][r])).x;
    }
    __builtin_amdgcn_s_waitcnt(0);
    {
      const int row = lane & 15;
      const int q0f = (lane >> 4) * 8;
      bf16x8 pa = *reinterpret_cast<const bf16x8*>(&p_lds[wave][row][q0f]);
#pragma unroll
      for (int t = 0; t < 8; ++t) {
        const int dcol = t * 16 + (lane & 15);
        bf16x8 bfrag = *reinterpret_cast<const bf16x8*>(
            &dot_lds[dcol][q0f ^ (dcol & 0x18)]);
        dv_acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            pa, bfrag, dv_acc[t], 0, 0, 0);
      }
    }

    __builtin_amdgcn_s_waitcnt(0);
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = (lane >> 4) * 4 + r;
#pragma unroll
      for (int nt = 0; nt < 2; ++nt)
        p_lds[wave][row][nt * 16 + col] =
            (short)__hip_bfloat16_raw(__float2bfloat16(dpt_acc[nt][r])).x;
    }
    __builtin_amdgcn_s_waitcnt(0);
    {
      const int row = lane & 15;
      const int q0f = (lane >> 4) * 8;
      bf16x8 pa = *reinterpret_cast<const bf16x8*>(&p_lds[wave][row][q0f]);
#pragma unroll
      for (int t = 0; t < 8; ++t) {
        const int dcol = t * 16 + (lane & 15);
        bf16x8 bfrag = *reinterpret_cast<const bf16x8*>(
            &qt_lds[dcol][q0f ^ (dcol & 0x18)]);
        dk_acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            pa, bfrag, dk_acc[t], 0, 0, 0);
      }
    }
  }

  if (!kv_active) return;
  const int col = lane & 15;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int kvrow = kv0_wave + (lane >> 4) * 4 + r;
#pragma unroll
    for (int t = 0; t < 8; ++t) {
      dKp[(long)kvrow * FA_D + t * 16 + col] = __float2bfloat16(dk_acc[t][r]);
      dVp[(long)kvrow * FA_D + t * 16 + col] = __float2bfloat16(dv_acc[t][r]);
    }
  }
}

extern "C" void fs_flash_attn_bwd(const void* q, const void* k, const void* v,
                                  const void* o, const void* dout,
                                  const float* lse, void* dq, void* dk,
                                  void* dv, float* delta_ws, int b, int h,
                                  int s, float scale, hipStream_t stream) {
  const long rows = (long)b * h * s;
  {
    long blocks = (rows + 3) / 4;
    if (blocks > 4096) blocks = 4096;
    hipLaunchKernelGGL(flash_delta_kernel, dim3((unsigned)blocks), dim3(256),
                       0, stream, (const bf16_t*)dout, (const bf16_t*)o,
                       delta_ws, rows);
  }
  dim3 grid((s + FA_QBLK - 1) / FA_QBLK, h, b);
  dim3 block(FA_WAVES * 64);
  hipLaunchKernelGGL(flash_attn_bwd_dq_kernel, grid, block, 0, stream,
                     (const bf16_t*)q, (const bf16_t*)k, (const bf16_t*)v,
                     (const bf16_t*)dout, lse, delta_ws, (bf16_t*)dq, b, h, s,
                     scale);
  hipLaunchKernelGGL(flash_attn_bwd_dkv_kernel, grid, block, 0, stream,
                     (const bf16_t*)q, (const bf16_t*)k, (const bf16_t*)v,
                     (const bf16_t*)dout, lse, delta_ws, (bf16_t*)dk,
                     (bf16_t*)dv, b, h, s, scale);
}
