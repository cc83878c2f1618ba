// Flash-style fused attention (fwd + FA2-style bwd) for MI355X (gfx950).
//
// Replaces the reference's flash_attn_cuda import (flash_attention.py:7) with
// a native CDNA4 kernel family: per Q-tile, stream K/V tiles through LDS,
// QK^T and PV on MFMA (v_mfma_f32_16x16x32_bf16), online softmax in fp32
// registers.  No SxS materialization (removes the reference's fused-softmax
// sk<=2048 cap and the 3x HBM round trip of the bmm path).
//
// Generality (round 2): kernels are templated on <int D, bool CAUSAL>:
//   D in {40, 64, 80, 96, 128, 160}  — covers LLaMA/GPT (128/96), BERT (64),
//       SD UNet self/cross attention (40/80/160).  D is padded to DP
//       (multiple of 32) in LDS/fragments; global loads are guarded at
//       8-element granularity (D % 8 == 0), pad lanes carry zeros.
//   CAUSAL=true : dense causal self-attention (sq == sk, sq % 64 == 0) —
//       byte-identical schedule to the round-1 kernel at D=128.
//   CAUSAL=false: bidirectional, supports cross-attention (sq != sk),
//       ragged sk (any length; tail masked), and per-batch key lengths
//       (klens[b], suffix padding masks a la BERT) — key col >= klen
//       contributes nothing to softmax, P, dK, dV.
//
// Structure (guide §B "fused attention prefill"):
//   block = 8 waves; wave w owns 16 q-rows => QBLK = 128; KVBLK = 64.
//   K staged [64][DP] with XOR swizzle (guide §6 G4 / T2: row-major bf16 is
//   a wide ds_read_b128 bank conflict; byte ^= (row&7)<<4 fixes it; LDS row
//   stride is padded to a power of two so the XOR stays a bijection).
//   V staged transposed [DP][64+pad] with kv-index swizzle so PV B-fragments
//   read 16B-contiguous kv runs.  async-STAGE split (T14): next tile's
//   global loads issue during current tile's MFMA.
// Outputs: O [b,h,sq,D] and LSE [b,h,sq] (softmax log-sum-exp, for bwd).

#include "common.h"

typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define FA_QBLK 128          // 8 waves x 16 q rows
#define FA_KVBLK 64
#define FA_WAVES 8
#define FA_VPAD 8
#define FB_T 32              // backward streaming tile

// padded head dim (fragment/LDS layout granularity)
constexpr int fa_dpad(int d) { return (d + 31) / 32 * 32; }
// LDS row stride in BYTES: power of two >= 2*DP so the (row&7)<<4 XOR
// swizzle is a bijection that never crosses rows
constexpr int fa_swb(int dp) {
  int need = 2 * dp, p = 128;
  while (p < need) p <<= 1;
  return p;
}

// XOR swizzle for row-major K/Q/V LDS tiles (16B-granular byte offsets)
__device__ __forceinline__ int kswz(int row, int byte_in_row) {
  return byte_in_row ^ ((row & 7) << 4);
}

// splitmix64 counter hash for attention dropout: fwd and bwd regenerate
// the SAME keep/drop decision per (bh, qrow, kcol) from (seed, index) —
// no mask tensor is ever materialized.
__device__ __forceinline__ unsigned int fa_hash(unsigned long long seed,
                                                unsigned long long idx) {
  unsigned long long z = seed + idx * 0x9E3779B97F4A7C15ull;
  z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ull;
  z = (z ^ (z >> 27)) * 0x94D049BB133111EBull;
  return (unsigned int)(z >> 32);
}

__device__ __forceinline__ short fa_bf16bits(float f) {
  return (short)__hip_bfloat16_raw(__float2bfloat16(f)).x;
}

// guarded 8-elem global load: zero vector beyond D (D % 8 == 0)
template <int D>
__device__ __forceinline__ bf16x8 fa_load8(const bf16_t* rowp, int d0) {
  if (d0 < D) return *reinterpret_cast<const bf16x8*>(rowp + d0);
  bf16x8 z;
#pragma unroll
  for (int j = 0; j < 8; ++j) z[j] = 0;
  return z;
}

// ===========================================================================
// FORWARD
// ===========================================================================
template <int D, bool CAUSAL>
__global__ __launch_bounds__(FA_WAVES * 64)
void flash_attn_fwd_kernel(const bf16_t* __restrict__ Q,
                           const bf16_t* __restrict__ K,
                           const bf16_t* __restrict__ V,
                           bf16_t* __restrict__ O,
                           float* __restrict__ LSE,
                           const int* __restrict__ klens,
                           int b, int h, int sq, int sk, float scale,
                           unsigned int drop_thresh, float keep_scale,
                           unsigned long long drop_seed) {
  constexpr int DP = fa_dpad(D);
  constexpr int NC = DP / 32;    // MFMA K-chunks
  constexpr int NTO = DP / 16;   // output d-tiles
  constexpr int SWB = fa_swb(DP);

  __shared__ char k_raw[FA_KVBLK * SWB];               // swizzled K rows
  __shared__ short vt_lds[DP][FA_KVBLK + FA_VPAD];     // V transposed
  __shared__ short p_lds[FA_WAVES][16][FA_KVBLK + FA_VPAD];

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int qb = blockIdx.x;
  const int head = blockIdx.y;
  const int batch = blockIdx.z;

  const bf16_t* Qp = Q + ((long)batch * h + head) * sq * D;
  const bf16_t* Kp = K + ((long)batch * h + head) * sk * D;
  const bf16_t* Vp = V + ((long)batch * h + head) * sk * D;
  bf16_t* Op = O + ((long)batch * h + head) * sq * D;

  const int klen = CAUSAL ? sk
                          : min(klens ? klens[batch] : sk, sk);

  const int q0 = qb * FA_QBLK + wave * 16;
  // partial blocks: OOB waves compute on clamped rows (must still hit every
  // __syncthreads) and skip their stores
  const bool q_active = q0 < sq;
  const int q0c = q_active ? q0 : (sq > 16 ? sq - 16 : 0);

  // ---- Q tile -> A-fragments (layout LA0: row=l%16, k=(l/16)*8+j;
  // verified on HW by scripts/mfma_probe), pre-scaled -----------------------
  bf16x8 q_frag[NC];
  {
    const int row = lane & 15;
    const int k0 = (lane >> 4) * 8;
    const int qrow = min(q0c + row, sq - 1);
#pragma unroll
    for (int c = 0; c < NC; ++c) {
      bf16x8 raw = fa_load8<D>(Qp + (long)qrow * D, c * 32 + k0);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        unsigned short u = (unsigned short)raw[j];
        float f = __uint_as_float(((unsigned int)u) << 16) * scale;
        raw[j] = fa_bf16bits(f);
      }
      q_frag[c] = raw;
    }
  }

  float m_run[4], l_run[4];
  f32x4 o_acc[NTO];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    m_run[r] = -INFINITY;
    l_run[r] = 0.f;
  }
#pragma unroll
  for (int t = 0; t < NTO; ++t) o_acc[t] = f32x4{0, 0, 0, 0};

  int n_kv_tiles;
  if (CAUSAL) {
    const int q_hi = min(qb * FA_QBLK + FA_QBLK, sq) - 1;
    n_kv_tiles = (q_hi / FA_KVBLK) + 1;
  } else {
    n_kv_tiles = (klen + FA_KVBLK - 1) / FA_KVBLK;
    if (n_kv_tiles < 1) n_kv_tiles = 1;  // empty rows still write O=0
  }

  // async-STAGE split (guide T14): global loads for tile t+1 are issued
  // DURING tile t's compute (registers k_reg/v_reg); only the cheap LDS
  // writes sit between the barriers — HBM latency hides under MFMA.
  const int tid = threadIdx.x;
  constexpr int ELEMS = FA_KVBLK * DP;          // elements per tile
  constexpr int SWEEPS = (ELEMS + FA_WAVES * 64 * 8 - 1) / (FA_WAVES * 64 * 8);
  bf16x8 k_reg[SWEEPS], v_reg[SWEEPS];
#pragma unroll
  for (int sweep = 0; sweep < SWEEPS; ++sweep) {
    const int i = tid * 8 + sweep * (FA_WAVES * 64 * 8);
    if (i < ELEMS) {
      const int kr = i / DP;
      const int kc = i % DP;
      const long krg = min(kr, sk - 1);
      k_reg[sweep] = fa_load8<D>(Kp + krg * D, kc);
      v_reg[sweep] = fa_load8<D>(Vp + krg * D, kc);
    }
  }

  for (int kt = 0; kt < n_kv_tiles; ++kt) {
    const int k_base = kt * FA_KVBLK;
    __syncthreads();
#pragma unroll
    for (int sweep = 0; sweep < SWEEPS; ++sweep) {
      const int i = tid * 8 + sweep * (FA_WAVES * 64 * 8);
      if (i < ELEMS) {
        const int kr = i / DP;
        const int kc = i % DP;
        *reinterpret_cast<bf16x8*>(k_raw + kr * SWB + kswz(kr, kc * 2)) =
            k_reg[sweep];
        bf16x8 vv = v_reg[sweep];
        // kv index XOR-swizzled by d bits 3-5 (the transpose scatter would
        // otherwise put 16 lanes of one K-row into ONE bank)
#pragma unroll
        for (int j = 0; j < 8; ++j)
          vt_lds[kc + j][kr ^ ((kc + j) & 0x38)] = vv[j];
      }
    }
    __syncthreads();
    if (kt + 1 < n_kv_tiles) {
      const int nb = (kt + 1) * FA_KVBLK;
#pragma unroll
      for (int sweep = 0; sweep < SWEEPS; ++sweep) {
        const int i = tid * 8 + sweep * (FA_WAVES * 64 * 8);
        if (i < ELEMS) {
          const int kr = i / DP;
          const int kc = i % DP;
          const long krg = min(nb + kr, sk - 1);
          k_reg[sweep] = fa_load8<D>(Kp + krg * D, kc);
          v_reg[sweep] = fa_load8<D>(Vp + krg * D, kc);
        }
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
        for (int c = 0; c < NC; ++c) {
          const int d0 = c * 32 + (lane >> 4) * 8;
          bf16x8 bfrag = *reinterpret_cast<const bf16x8*>(
              k_raw + krow * SWB + kswz(krow, d0 * 2));
          s_acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              q_frag[c], bfrag, s_acc[nt], 0, 0, 0);
        }
      }
    }

    // ---- mask + online softmax -------------------------------------------
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
        if (CAUSAL ? (kcol > qrow) : (kcol >= klen)) v = -INFINITY;
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
    bool need_rescale = false;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const float m_new = fmaxf(m_run[r], tile_max[r]);
      alpha[r] = (m_run[r] == -INFINITY) ? 0.f : __expf(m_run[r] - m_new);
      if (m_new != m_run[r]) need_rescale = true;
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
    // attention dropout on P (normalizer l uses the UNdropped sum, like
    // eager dropout(softmax(S))); uniform branch, free when p == 0
    if (drop_thresh) {
      const unsigned long long bh_base =
          ((unsigned long long)batch * h + head) * sq;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const unsigned long long qi =
            (bh_base + (unsigned long long)(q0c + (lane >> 4) * 4 + r)) *
            (unsigned long long)sk;
#pragma unroll
        for (int nt = 0; nt < 4; ++nt) {
          const int kcol = k_base + nt * 16 + col;
          p[nt][r] = (fa_hash(drop_seed, qi + kcol) < drop_thresh)
                         ? 0.f : p[nt][r] * keep_scale;
        }
      }
    }
    // skip the O-wide rescale when no row max moved (defer-max lite, T13)
    if (__any(need_rescale)) {
#pragma unroll
      for (int t = 0; t < NTO; ++t) {
#pragma unroll
        for (int r = 0; r < 4; ++r) o_acc[t][r] *= alpha[r];
      }
    }

    // ---- P -> per-wave LDS, then PV ---------------------------------------
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = (lane >> 4) * 4 + r;
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
        p_lds[wave][row][nt * 16 + col] = fa_bf16bits(p[nt][r]);
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
      for (int t = 0; t < NTO; ++t) {
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
    if (qrow >= sq) continue;
    const float inv_l = (l_run[r] > 0.f) ? 1.f / l_run[r] : 0.f;
#pragma unroll
    for (int t = 0; t < NTO; ++t) {
      const int d = t * 16 + col;
      if (d < D) Op[(long)qrow * D + d] = __float2bfloat16(o_acc[t][r] * inv_l);
    }
    if (col == 0 && LSE) {
      LSE[((long)batch * h + head) * sq + qrow] =
          m_run[r] + logf(fmaxf(l_run[r], 1e-30f));
    }
  }
}

// ===========================================================================
// BACKWARD
// ===========================================================================
// FlashAttention-2-style: delta = rowsum(dO*O); then
//   dQ kernel (q-major):   S=(sQ)K^T; P=exp(S-LSE); dP=dO V^T;
//                          dS=P*(dP-delta)*s; dQ += dS K
//   dKdV kernel (kv-major): St=(sK)Q^T; Pt=exp(St-LSE[col]); dPt=V dO^T;
//                          dV += Pt dO; dSt=Pt*(dPt-delta[col])*s; dK += dSt Q
// Same fragment layouts / swizzles as forward.  Streaming tiles are 32-wide
// (FB_T) so LDS stays small -> 3-4 blocks/CU.

template <int D>
__global__ __launch_bounds__(256)
void flash_delta_kernel(const bf16_t* __restrict__ dO,
                        const bf16_t* __restrict__ O,
                        float* __restrict__ delta, long rows) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const long row0 = (long)blockIdx.x * 4 + wid;
  const long stride = (long)gridDim.x * 4;
  for (long row = row0; row < rows; row += stride) {
    const bf16_t* dop = dO + row * D;
    const bf16_t* op = O + row * D;
    float sacc = 0.f;
    for (int i = lane * 2; i < D; i += 128) {
#pragma unroll
      for (int j = 0; j < 2; ++j)
        sacc += __bfloat162float(dop[i + j]) * __bfloat162float(op[i + j]);
    }
    sacc = wave_reduce_sum(sacc);
    if (lane == 0) delta[row] = sacc;
  }
}

// Q-major: each block owns 128 q rows (wave -> 16), streams KV in 32-tiles.
template <int D, bool CAUSAL>
__global__ __launch_bounds__(FA_WAVES * 64)
void flash_attn_bwd_dq_kernel(const bf16_t* __restrict__ Q,
                              const bf16_t* __restrict__ K,
                              const bf16_t* __restrict__ V,
                              const bf16_t* __restrict__ dO,
                              const float* __restrict__ LSE,
                              const float* __restrict__ Delta,
                              bf16_t* __restrict__ dQ,
                              const int* __restrict__ klens,
                              int b, int h, int sq, int sk, float scale,
                              unsigned int drop_thresh, float keep_scale,
                              unsigned long long drop_seed) {
  constexpr int DP = fa_dpad(D);
  constexpr int NC = DP / 32;
  constexpr int NTO = DP / 16;
  constexpr int SWB = fa_swb(DP);

  __shared__ char k_raw[FB_T * SWB];                  // K rows, swizzled
  __shared__ char v_raw[FB_T * SWB];                  // V rows, swizzled
  __shared__ short kt_lds[DP][FB_T + FA_VPAD];        // K^T, kv-swizzled
  __shared__ short p_lds[FA_WAVES][16][FB_T + FA_VPAD];

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const long bh = (long)blockIdx.z * h + blockIdx.y;
  const bf16_t* Qp = Q + bh * sq * D;
  const bf16_t* Kp = K + bh * sk * D;
  const bf16_t* Vp = V + bh * sk * D;
  const bf16_t* dOp = dO + bh * sq * D;
  bf16_t* dQp = dQ + bh * sq * D;
  const float* lse = LSE + bh * sq;
  const float* dlt = Delta + bh * sq;

  const int klen = CAUSAL ? sk
                          : min(klens ? klens[blockIdx.z] : sk, sk);

  const int q0 = blockIdx.x * FA_QBLK + wave * 16;
  const bool q_active = q0 < sq;
  const int q0c = q_active ? q0 : (sq > 16 ? sq - 16 : 0);

  bf16x8 q_frag[NC], do_frag[NC];
  {
    const int row = lane & 15;
    const int k0 = (lane >> 4) * 8;
    const int qrow = min(q0c + row, sq - 1);
#pragma unroll
    for (int c = 0; c < NC; ++c) {
      bf16x8 raw = fa_load8<D>(Qp + (long)qrow * D, c * 32 + k0);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        unsigned short u = (unsigned short)raw[j];
        float f = __uint_as_float(((unsigned int)u) << 16) * scale;
        raw[j] = fa_bf16bits(f);
      }
      q_frag[c] = raw;
      do_frag[c] = fa_load8<D>(dOp + (long)qrow * D, c * 32 + k0);
    }
  }
  float lse_r[4], dlt_r[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int qrow = min(q0c + (lane >> 4) * 4 + r, sq - 1);
    lse_r[r] = lse[qrow];
    dlt_r[r] = dlt[qrow];
  }

  f32x4 dq_acc[NTO];
#pragma unroll
  for (int t = 0; t < NTO; ++t) dq_acc[t] = f32x4{0, 0, 0, 0};

  int n_kv_tiles;
  if (CAUSAL) {
    const int q_hi = min(blockIdx.x * FA_QBLK + FA_QBLK, sq) - 1;
    n_kv_tiles = (q_hi / FB_T) + 1;
  } else {
    n_kv_tiles = (klen + FB_T - 1) / FB_T;
  }

  // async-STAGE (T14, like the forward): tile t+1 K/V global loads are
  // issued DURING tile t's MFMA so HBM latency hides under compute; only
  // the LDS writes sit between the barriers.
  constexpr int BELEMS = FB_T * DP;
  constexpr int BSWEEPS = (BELEMS + FA_WAVES * 64 * 8 - 1)
                          / (FA_WAVES * 64 * 8);
  bf16x8 kst[BSWEEPS], vst[BSWEEPS];
  {
    const int tid = threadIdx.x;
    const int kt0_base = (CAUSAL ? 0 : 0) * FB_T;
#pragma unroll
    for (int sw = 0; sw < BSWEEPS; ++sw) {
      const int i = tid * 8 + sw * (FA_WAVES * 64 * 8);
      if (i < BELEMS) {
        const int kr = i / DP;
        const int kc = i % DP;
        const long krg = min(kt0_base + kr, sk - 1);
        kst[sw] = fa_load8<D>(Kp + krg * D, kc);
        vst[sw] = fa_load8<D>(Vp + krg * D, kc);
      }
    }
  }
  for (int kt = 0; kt < n_kv_tiles; ++kt) {
    __syncthreads();
    {
      const int tid = threadIdx.x;
#pragma unroll
      for (int sw = 0; sw < BSWEEPS; ++sw) {
        const int i = tid * 8 + sw * (FA_WAVES * 64 * 8);
        if (i < BELEMS) {
          const int kr = i / DP;
          const int kc = i % DP;
          bf16x8 kk = kst[sw];
          *reinterpret_cast<bf16x8*>(
              k_raw + kr * SWB + kswz(kr, kc * 2)) = kk;
          *reinterpret_cast<bf16x8*>(
              v_raw + kr * SWB + kswz(kr, kc * 2)) = vst[sw];
#pragma unroll
          for (int j = 0; j < 8; ++j)
            kt_lds[kc + j][kr ^ ((kc + j) & 0x18)] = kk[j];
        }
      }
    }
    __syncthreads();
    const int k_base = kt * FB_T;
    if (kt + 1 < n_kv_tiles) {
      const int nb = (kt + 1) * FB_T;
      const int tid = threadIdx.x;
#pragma unroll
      for (int sw = 0; sw < BSWEEPS; ++sw) {
        const int i = tid * 8 + sw * (FA_WAVES * 64 * 8);
        if (i < BELEMS) {
          const int kr = i / DP;
          const int kc = i % DP;
          const long krg = min(nb + kr, sk - 1);
          kst[sw] = fa_load8<D>(Kp + krg * D, kc);
          vst[sw] = fa_load8<D>(Vp + krg * D, kc);
        }
      }
    }

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
        for (int c = 0; c < NC; ++c) {
          const int d0 = c * 32 + (lane >> 4) * 8;
          bf16x8 kb = *reinterpret_cast<const bf16x8*>(
              k_raw + krow * SWB + kswz(krow, d0 * 2));
          s_acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              q_frag[c], kb, s_acc[nt], 0, 0, 0);
          bf16x8 vb = *reinterpret_cast<const bf16x8*>(
              v_raw + krow * SWB + kswz(krow, d0 * 2));
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
        const bool masked = CAUSAL ? (kcol > qrow) : (kcol >= klen);
        float pv = masked ? 0.f : __expf(s_acc[nt][r] - lse_r[r]);
        float dpv = dp_acc[nt][r];
        if (drop_thresh) {
          const unsigned long long idx =
              (bh * (unsigned long long)sq + qrow) *
                  (unsigned long long)sk + kcol;
          dpv = (fa_hash(drop_seed, idx) < drop_thresh)
                    ? 0.f : dpv * keep_scale;
        }
        float ds = pv * (dpv - dlt_r[r]) * scale;
        p_lds[wave][row][nt * 16 + col] = fa_bf16bits(ds);
      }
    }
    __builtin_amdgcn_s_waitcnt(0);

    {
      const int row = lane & 15;
      const int kv0 = (lane >> 4) * 8;
      bf16x8 pa = *reinterpret_cast<const bf16x8*>(&p_lds[wave][row][kv0]);
#pragma unroll
      for (int t = 0; t < NTO; ++t) {
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
    if (qrow >= sq) continue;
#pragma unroll
    for (int t = 0; t < NTO; ++t) {
      const int d = t * 16 + col;
      if (d < D) dQp[(long)qrow * D + d] = __float2bfloat16(dq_acc[t][r]);
    }
  }
}

// KV-major: each block owns 128 kv rows (wave -> 16), streams Q in 32-tiles.
template <int D, bool CAUSAL>
__global__ __launch_bounds__(FA_WAVES * 64)
void flash_attn_bwd_dkv_kernel(const bf16_t* __restrict__ Q,
                               const bf16_t* __restrict__ K,
                               const bf16_t* __restrict__ V,
                               const bf16_t* __restrict__ dO,
                               const float* __restrict__ LSE,
                               const float* __restrict__ Delta,
                               bf16_t* __restrict__ dK,
                               bf16_t* __restrict__ dV,
                               const int* __restrict__ klens,
                               int b, int h, int sq, int sk, float scale,
                               unsigned int drop_thresh, float keep_scale,
                               unsigned long long drop_seed) {
  constexpr int DP = fa_dpad(D);
  constexpr int NC = DP / 32;
  constexpr int NTO = DP / 16;
  constexpr int SWB = fa_swb(DP);

  __shared__ char q_raw[FB_T * SWB];
  __shared__ char do_raw[FB_T * SWB];
  __shared__ short qt_lds[DP][FB_T + FA_VPAD];
  __shared__ short dot_lds[DP][FB_T + FA_VPAD];
  __shared__ short p_lds[FA_WAVES][16][FB_T + FA_VPAD];

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const long bh = (long)blockIdx.z * h + blockIdx.y;
  const bf16_t* Qp = Q + bh * sq * D;
  const bf16_t* Kp = K + bh * sk * D;
  const bf16_t* Vp = V + bh * sk * D;
  const bf16_t* dOp = dO + bh * sq * D;
  bf16_t* dKp = dK + bh * sk * D;
  bf16_t* dVp = dV + bh * sk * D;
  const float* lse = LSE + bh * sq;
  const float* dlt = Delta + bh * sq;

  const int klen = CAUSAL ? sk
                          : min(klens ? klens[blockIdx.z] : sk, sk);

  // each block owns FA_WAVES*16 kv rows
  const int kv0_blk = blockIdx.x * (FA_WAVES * 16);
  const int kv0_wave = kv0_blk + wave * 16;
  const bool kv_active = kv0_wave < sk;
  const int kv0c = kv_active ? kv0_wave : (sk > 16 ? sk - 16 : 0);

  bf16x8 k_frag[NC], v_frag[NC];
  {
    const int row = lane & 15;
    const int c0 = (lane >> 4) * 8;
    const int kvrow = min(kv0c + row, sk - 1);
#pragma unroll
    for (int c = 0; c < NC; ++c) {
      bf16x8 raw = fa_load8<D>(Kp + (long)kvrow * D, c * 32 + c0);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        unsigned short u = (unsigned short)raw[j];
        float f = __uint_as_float(((unsigned int)u) << 16) * scale;
        raw[j] = fa_bf16bits(f);
      }
      k_frag[c] = raw;
      v_frag[c] = fa_load8<D>(Vp + (long)kvrow * D, c * 32 + c0);
    }
  }

  f32x4 dv_acc[NTO], dk_acc[NTO];
#pragma unroll
  for (int t = 0; t < NTO; ++t) {
    dv_acc[t] = f32x4{0, 0, 0, 0};
    dk_acc[t] = f32x4{0, 0, 0, 0};
  }

  const int first_qt = CAUSAL ? kv0_blk / FB_T : 0;
  const int n_q_tiles = (sq + FB_T - 1) / FB_T;

  constexpr int BELEMS = FB_T * DP;
  constexpr int BSWEEPS = (BELEMS + FA_WAVES * 64 * 8 - 1)
                          / (FA_WAVES * 64 * 8);
  bf16x8 qst[BSWEEPS], dst[BSWEEPS];
  {
    const int tid = threadIdx.x;
    const int q0_base = first_qt * FB_T;
#pragma unroll
    for (int sw = 0; sw < BSWEEPS; ++sw) {
      const int i = tid * 8 + sw * (FA_WAVES * 64 * 8);
      if (i < BELEMS) {
        const int qr = i / DP;
        const int qc = i % DP;
        const long qrg = min(q0_base + qr, sq - 1);
        qst[sw] = fa_load8<D>(Qp + qrg * D, qc);
        dst[sw] = fa_load8<D>(dOp + qrg * D, qc);
      }
    }
  }
  for (int qt = first_qt; qt < n_q_tiles; ++qt) {
    const int q_base = qt * FB_T;
    __syncthreads();
    {
      const int tid = threadIdx.x;
#pragma unroll
      for (int sw = 0; sw < BSWEEPS; ++sw) {
        const int i = tid * 8 + sw * (FA_WAVES * 64 * 8);
        if (i < BELEMS) {
          const int qr = i / DP;
          const int qc = i % DP;
          bf16x8 qq = qst[sw];
          bf16x8 dd = dst[sw];
          *reinterpret_cast<bf16x8*>(
              q_raw + qr * SWB + kswz(qr, qc * 2)) = qq;
          *reinterpret_cast<bf16x8*>(
              do_raw + qr * SWB + kswz(qr, qc * 2)) = dd;
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            qt_lds[qc + j][qr ^ ((qc + j) & 0x18)] = qq[j];
            dot_lds[qc + j][qr ^ ((qc + j) & 0x18)] = dd[j];
          }
        }
      }
    }
    __syncthreads();
    if (qt + 1 < n_q_tiles) {
      const int nb = (qt + 1) * FB_T;
      const int tid = threadIdx.x;
#pragma unroll
      for (int sw = 0; sw < BSWEEPS; ++sw) {
        const int i = tid * 8 + sw * (FA_WAVES * 64 * 8);
        if (i < BELEMS) {
          const int qr = i / DP;
          const int qc = i % DP;
          const long qrg = min(nb + qr, sq - 1);
          qst[sw] = fa_load8<D>(Qp + qrg * D, qc);
          dst[sw] = fa_load8<D>(dOp + qrg * D, qc);
        }
      }
    }

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
        for (int c = 0; c < NC; ++c) {
          const int d0 = c * 32 + (lane >> 4) * 8;
          bf16x8 qb = *reinterpret_cast<const bf16x8*>(
              q_raw + qrow * SWB + kswz(qrow, d0 * 2));
          st_acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              k_frag[c], qb, st_acc[nt], 0, 0, 0);
          bf16x8 db = *reinterpret_cast<const bf16x8*>(
              do_raw + qrow * SWB + kswz(qrow, d0 * 2));
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
      const int qcolc = min(qcol, sq - 1);
      const float lse_c = lse[qcolc];
      const float dlt_c = dlt[qcolc];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int kvrow = kv0c + (lane >> 4) * 4 + r;
        bool masked = CAUSAL ? (qcol < kvrow)
                             : (kvrow >= klen || qcol >= sq);
        float pv = masked ? 0.f : __expf(st_acc[nt][r] - lse_c);
        float dm = 1.f;
        if (drop_thresh) {
          const unsigned long long idx =
              (bh * (unsigned long long)sq + qcol) *
                  (unsigned long long)sk + kvrow;
          dm = (fa_hash(drop_seed, idx) < drop_thresh) ? 0.f : keep_scale;
        }
        pt[nt][r] = pv * dm;                       // dropped P feeds dV
        dpt_acc[nt][r] = pv * (dm * dpt_acc[nt][r] - dlt_c) * scale;  // dSt
      }
    }

#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = (lane >> 4) * 4 + r;
#pragma unroll
      for (int nt = 0; nt < 2; ++nt)
        p_lds[wave][row][nt * 16 + col] = fa_bf16bits(pt[nt][r]);
    }
    __builtin_amdgcn_s_waitcnt(0);
    {
      const int row = lane & 15;
      const int q0f = (lane >> 4) * 8;
      bf16x8 pa = *reinterpret_cast<const bf16x8*>(&p_lds[wave][row][q0f]);
#pragma unroll
      for (int t = 0; t < NTO; ++t) {
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
        p_lds[wave][row][nt * 16 + col] = fa_bf16bits(dpt_acc[nt][r]);
    }
    __builtin_amdgcn_s_waitcnt(0);
    {
      const int row = lane & 15;
      const int q0f = (lane >> 4) * 8;
      bf16x8 pa = *reinterpret_cast<const bf16x8*>(&p_lds[wave][row][q0f]);
#pragma unroll
      for (int t = 0; t < NTO; ++t) {
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
    if (kvrow >= sk) continue;
#pragma unroll
    for (int t = 0; t < NTO; ++t) {
      const int d = t * 16 + col;
      if (d < D) {
        dKp[(long)kvrow * D + d] = __float2bfloat16(dk_acc[t][r]);
        dVp[(long)kvrow * D + d] = __float2bfloat16(dv_acc[t][r]);
      }
    }
  }
}

// ===========================================================================
// launchers
// ===========================================================================
template <int D>
static void launch_fwd(bool causal, const void* q, const void* k,
                       const void* v, void* o, float* lse, const int* klens,
                       int b, int h, int sq, int sk, float scale,
                       unsigned int dth, float ksc, unsigned long long seed,
                       hipStream_t stream) {
  dim3 grid((sq + FA_QBLK - 1) / FA_QBLK, h, b);
  dim3 block(FA_WAVES * 64);
  if (causal)
    hipLaunchKernelGGL((flash_attn_fwd_kernel<D, true>), grid, block, 0,
                       stream, (const bf16_t*)q, (const bf16_t*)k,
                       (const bf16_t*)v, (bf16_t*)o, lse, klens, b, h, sq, sk,
                       scale, dth, ksc, seed);
  else
    hipLaunchKernelGGL((flash_attn_fwd_kernel<D, false>), grid, block, 0,
                       stream, (const bf16_t*)q, (const bf16_t*)k,
                       (const bf16_t*)v, (bf16_t*)o, lse, klens, b, h, sq, sk,
                       scale, dth, ksc, seed);
}

extern "C" void fs_flash_attn_fwd(const void* q, const void* k, const void* v,
                                  void* o, float* lse, const int* klens,
                                  int b, int h, int sq, int sk, int d,
                                  int causal, float scale, float drop_p,
                                  unsigned long long seed,
                                  hipStream_t stream) {
  const unsigned int dth =
      (drop_p > 0.f) ? (unsigned int)(drop_p * 4294967296.0) : 0u;
  const float ksc = (drop_p > 0.f) ? 1.f / (1.f - drop_p) : 1.f;
#define FWD_CASE(DD) case DD: launch_fwd<DD>(causal, q, k, v, o, lse, klens, \
    b, h, sq, sk, scale, dth, ksc, seed, stream); break;
  switch (d) {
    FWD_CASE(40) FWD_CASE(64) FWD_CASE(80)
    FWD_CASE(96) FWD_CASE(128) FWD_CASE(160)
  }
#undef FWD_CASE
}

template <int D>
static void launch_bwd(bool causal, const void* q, const void* k,
                       const void* v, const void* o, const void* dout,
                       const float* lse, void* dq, void* dk, void* dv,
                       float* delta_ws, const int* klens, int b, int h,
                       int sq, int sk, float scale, unsigned int dth,
                       float ksc, unsigned long long seed,
                       hipStream_t stream) {
  const long rows = (long)b * h * sq;
  {
    long blocks = (rows + 3) / 4;
    if (blocks > 4096) blocks = 4096;
    hipLaunchKernelGGL((flash_delta_kernel<D>), dim3((unsigned)blocks),
                       dim3(256), 0, stream, (const bf16_t*)dout,
                       (const bf16_t*)o, delta_ws, rows);
  }
  dim3 gq((sq + FA_QBLK - 1) / FA_QBLK, h, b);
  dim3 gkv((sk + FA_WAVES * 16 - 1) / (FA_WAVES * 16), h, b);
  dim3 block(FA_WAVES * 64);
  if (causal) {
    hipLaunchKernelGGL((flash_attn_bwd_dq_kernel<D, true>), gq, block, 0,
                       stream, (const bf16_t*)q, (const bf16_t*)k,
                       (const bf16_t*)v, (const bf16_t*)dout, lse, delta_ws,
                       (bf16_t*)dq, klens, b, h, sq, sk, scale, dth, ksc,
                       seed);
    hipLaunchKernelGGL((flash_attn_bwd_dkv_kernel<D, true>), gkv, block, 0,
                       stream, (const bf16_t*)q, (const bf16_t*)k,
                       (const bf16_t*)v, (const bf16_t*)dout, lse, delta_ws,
                       (bf16_t*)dk, (bf16_t*)dv, klens, b, h, sq, sk, scale,
                       dth, ksc, seed);
  } else {
    hipLaunchKernelGGL((flash_attn_bwd_dq_kernel<D, false>), gq, block, 0,
                       stream, (const bf16_t*)q, (const bf16_t*)k,
                       (const bf16_t*)v, (const bf16_t*)dout, lse, delta_ws,
                       (bf16_t*)dq, klens, b, h, sq, sk, scale, dth, ksc,
                       seed);
    hipLaunchKernelGGL((flash_attn_bwd_dkv_kernel<D, false>), gkv, block, 0,
                       stream, (const bf16_t*)q, (const bf16_t*)k,
                       (const bf16_t*)v, (const bf16_t*)dout, lse, delta_ws,
                       (bf16_t*)dk, (bf16_t*)dv, klens, b, h, sq, sk, scale,
                       dth, ksc, seed);
  }
}

extern "C" void fs_flash_attn_bwd(const void* q, const void* k, const void* v,
                                  const void* o, const void* dout,
                                  const float* lse, void* dq, void* dk,
                                  void* dv, float* delta_ws, const int* klens,
                                  int b, int h, int sq, int sk, int d,
                                  int causal, float scale, float drop_p,
                                  unsigned long long seed,
                                  hipStream_t stream) {
  const unsigned int dth =
      (drop_p > 0.f) ? (unsigned int)(drop_p * 4294967296.0) : 0u;
  const float ksc = (drop_p > 0.f) ? 1.f / (1.f - drop_p) : 1.f;
#define BWD_CASE(DD) case DD: launch_bwd<DD>(causal, q, k, v, o, dout, lse, \
    dq, dk, dv, delta_ws, klens, b, h, sq, sk, scale, dth, ksc, seed, \
    stream); break;
  switch (d) {
    BWD_CASE(40) BWD_CASE(64) BWD_CASE(80)
    BWD_CASE(96) BWD_CASE(128) BWD_CASE(160)
  }
#undef BWD_CASE
}

// ===========================================================================
// FORWARD v3 — 8-wave swapped-QK^T schedule (guide §B 8-warp ladder)
// ===========================================================================
// Causal, D=128 (the flagship LLaMA shape).  Differences vs the general
// kernel above:
//   - wave owns 32 q rows (QBLK = 256/block): K/V LDS traffic is amortized
//     over 2x more query rows;
//   - mfma_f32_32x32x16_bf16 with SWAPPED operands (A = K-tile, B = Q):
//     each lane holds a full 32-wide P-row slice for ONE q row in
//     registers, so the row max/sum are 31 in-lane ops + one cross-half
//     shuffle — no 4-round shuffle-reduce trees, and softmax state m/l is
//     a per-lane scalar;
//   - P crosses to the PV A-operand through a per-wave LDS exchange
//     (bf16, 8B stores / 16B reads).
// Layouts verified on HW by scripts/mfma32_probe.hip.

#define V3_QBLK 256          // 8 waves x 32 q rows
#define V3_KVBLK 64

typedef __attribute__((ext_vector_type(16))) float f32x16;

template <int D, bool CAUSAL, bool HAS_DROP>
__global__ __launch_bounds__(FA_WAVES * 64)
void flash_attn_fwd_v3_kernel(const bf16_t* __restrict__ Q,
                              const bf16_t* __restrict__ K,
                              const bf16_t* __restrict__ V,
                              bf16_t* __restrict__ O,
                              float* __restrict__ LSE,
                              const int* __restrict__ klens,
                              int b, int h, int s, float scale,
                              unsigned int drop_thresh, float keep_scale,
                              unsigned long long drop_seed) {
  constexpr int NC = D / 16;     // MFMA K=16 chunks
  constexpr int NTO = D / 32;    // 32-wide output d-tiles
  constexpr int SWB = fa_swb(D);

  __shared__ char k_raw[2][V3_KVBLK * SWB];             // swizzled K rows
  __shared__ short vt_lds[2][D][V3_KVBLK + FA_VPAD];    // V transposed
  // (double-buffered: tile t+1 stages into buf^1 while buf holds tile t,
  // so the loop needs ONE barrier per tile instead of two.
  // P stays fully in registers via cvt-pack + permlane32_swap — T12;
  // alpha/1-l broadcasts ride lane shuffles: no LDS exchange, no
  // wave-wide lgkmcnt drains in the softmax/PV region)

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int hi = lane >> 5;      // lane half
  const int ln = lane & 31;
  const long bh = (long)blockIdx.z * h + blockIdx.y;
  const bf16_t* Qp = Q + bh * s * D;
  const bf16_t* Kp = K + bh * s * D;
  const bf16_t* Vp = V + bh * s * D;
  bf16_t* Op = O + bh * s * D;

  const int klen = CAUSAL ? s : min(klens ? klens[blockIdx.z] : s, s);

  const int q0w = blockIdx.x * V3_QBLK + wave * 32;
  const bool q_active = q0w < s;
  const int q0c = q_active ? q0w : s - 32;
  const int my_q = q0c + ln;      // softmax row this lane owns

  // ---- Q -> B-fragments (K=16 chunks), pre-scaled -----------------------
  bf16x8 q_frag[NC];
#pragma unroll
  for (int c = 0; c < NC; ++c) {
    bf16x8 raw = *reinterpret_cast<const bf16x8*>(
        Qp + (long)my_q * D + c * 16 + hi * 8);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      unsigned short u = (unsigned short)raw[j];
      float f = __uint_as_float(((unsigned int)u) << 16) * scale;
      raw[j] = fa_bf16bits(f);
    }
    q_frag[c] = raw;
  }

  float m_run = -INFINITY, l_run = 0.f;   // per-lane: row my_q
  f32x16 o_acc[NTO];
#pragma unroll
  for (int t = 0; t < NTO; ++t) {
#pragma unroll
    for (int r = 0; r < 16; ++r) o_acc[t][r] = 0.f;
  }

  int n_kv_tiles;
  if (CAUSAL) {
    const int q_hi_row = min(blockIdx.x * V3_QBLK + V3_QBLK, s) - 1;
    n_kv_tiles = (q_hi_row / V3_KVBLK) + 1;
  } else {
    n_kv_tiles = (klen + V3_KVBLK - 1) / V3_KVBLK;
    if (n_kv_tiles < 1) n_kv_tiles = 1;
  }

  // async-STAGE staging registers
  const int tid = threadIdx.x;
  constexpr int ELEMS = V3_KVBLK * D;
  constexpr int SWEEPS = (ELEMS + FA_WAVES * 64 * 8 - 1) / (FA_WAVES * 64 * 8);
  bf16x8 k_reg[SWEEPS], v_reg[SWEEPS];
#pragma unroll
  for (int sweep = 0; sweep < SWEEPS; ++sweep) {
    const int i = tid * 8 + sweep * (FA_WAVES * 64 * 8);
    if (i < ELEMS) {
      const long kr = min(i / D, s - 1);
      const int kc = i % D;
      k_reg[sweep] = *reinterpret_cast<const bf16x8*>(Kp + kr * D + kc);
      v_reg[sweep] = *reinterpret_cast<const bf16x8*>(Vp + kr * D + kc);
    }
  }

  // stage tile 0 into buffer 0
  {
#pragma unroll
    for (int sweep = 0; sweep < SWEEPS; ++sweep) {
      const int i = tid * 8 + sweep * (FA_WAVES * 64 * 8);
      if (i < ELEMS) {
        const int kr = i / D;
        const int kc = i % D;
        *reinterpret_cast<bf16x8*>(
            k_raw[0] + kr * SWB + kswz(kr, kc * 2)) = k_reg[sweep];
        bf16x8 vv = v_reg[sweep];
#pragma unroll
        for (int j = 0; j < 8; ++j)
          vt_lds[0][kc + j][kr ^ ((kc + j) & 0x38)] = vv[j];
      }
    }
  }
  int cur = 0;
  for (int kt = 0; kt < n_kv_tiles; ++kt) {
    const int k_base = kt * V3_KVBLK;
    __syncthreads();  // publishes buf[cur]; everyone done with buf[cur^1]
    if (kt + 1 < n_kv_tiles) {
      const int nb = (kt + 1) * V3_KVBLK;
#pragma unroll
      for (int sweep = 0; sweep < SWEEPS; ++sweep) {
        const int i = tid * 8 + sweep * (FA_WAVES * 64 * 8);
        if (i < ELEMS) {
          const long kr = min((long)(nb + i / D), (long)s - 1);
          const int kc = i % D;
          k_reg[sweep] = *reinterpret_cast<const bf16x8*>(Kp + kr * D + kc);
          v_reg[sweep] = *reinterpret_cast<const bf16x8*>(Vp + kr * D + kc);
        }
      }
    }

    // ---- S^T = K_tile @ (sQ)^T : two 32x32 kv-tiles ---------------------
    f32x16 st[2];
#pragma unroll
    for (int nt = 0; nt < 2; ++nt) {
#pragma unroll
      for (int r = 0; r < 16; ++r) st[nt][r] = 0.f;
    }
#pragma unroll
    for (int nt = 0; nt < 2; ++nt) {
      const int krow = nt * 32 + ln;
#pragma unroll
      for (int c = 0; c < NC; ++c) {
        bf16x8 a = *reinterpret_cast<const bf16x8*>(
            k_raw[cur] + krow * SWB + kswz(krow, (c * 16 + hi * 8) * 2));
        st[nt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, q_frag[c],
                                                         st[nt], 0, 0, 0);
      }
    }
    // lane now holds S^T[kv][q = my row]: kv = k_base + nt*32 + crow(r,hi)
    // with crow(r,hi) = (r&3) + 8*(r>>2) + 4*hi

    // ---- causal mask + in-register online softmax ----------------------
    float p[32];
    float tmax = -INFINITY;
#pragma unroll
    for (int nt = 0; nt < 2; ++nt) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int kcol = k_base + nt * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
        float v = st[nt][r];
        if (CAUSAL ? (kcol > my_q) : (kcol >= klen)) v = -INFINITY;
        p[nt * 16 + r] = v;
        tmax = fmaxf(tmax, v);
      }
    }
    tmax = fmaxf(tmax, __shfl_xor(tmax, 32, 64));  // combine lane halves
    const float m_new = fmaxf(m_run, tmax);
    const float alpha = (m_run == -INFINITY) ? 0.f : __expf(m_run - m_new);
    const bool rescale = (m_new != m_run);
    m_run = m_new;
    float rs = 0.f;
#pragma unroll
    for (int i = 0; i < 32; ++i) {
      float e = (p[i] == -INFINITY) ? 0.f : __expf(p[i] - m_new);
      p[i] = e;
      rs += e;
    }
    rs += __shfl_xor(rs, 32, 64);
    l_run = l_run * alpha + rs;

    // attention dropout on P (normalizer keeps the undropped sum)
    if (HAS_DROP && drop_thresh) {
      const unsigned long long qi =
          (bh * (unsigned long long)s + my_q) * (unsigned long long)s;
#pragma unroll
      for (int nt = 0; nt < 2; ++nt) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int kcol =
              k_base + nt * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
          float& pv = p[nt * 16 + r];
          pv = (fa_hash(drop_seed, qi + kcol) < drop_thresh)
                   ? 0.f : pv * keep_scale;
        }
      }
    }

    // rescale: alpha for o-row crow(r,hi) fetched by lane shuffle
    // (alpha is per-q-row, identical in lanes l and l+32)
    if (__any(rescale)) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const float al = __shfl(alpha, (r & 3) + 8 * (r >> 2) + 4 * hi, 64);
#pragma unroll
        for (int t = 0; t < NTO; ++t) o_acc[t][r] *= al;
      }
    }

    // ---- P -> A-fragments in registers (T12) ----------------------------
    // Lane layout: p holds, for its own q row, kv octs G = 4*nt + g with
    // 4-runs at kv = 8G + 4*hi + j.  A-frag for PV chunk c2 needs kv
    // 16c2 + 8*hi' + j: word pair k from oct 2c2 (hi=0 half) and oct
    // 2c2+1 (hi=1 half).  One permlane32_swap per word yields BOTH
    // halves' targets: out0 = words 0/1, out1 = words 2/3, uniformly.
    unsigned int pw[8][2];  // [oct][word]: packed bf16 pairs of own run
#pragma unroll
    for (int G = 0; G < 8; ++G) {
      const int base = (G >> 2) * 16 + (G & 3) * 4;
#pragma unroll
      for (int k = 0; k < 2; ++k) {
        const unsigned int lo =
            (unsigned short)fa_bf16bits(p[base + 2 * k]);
        const unsigned int hi_b =
            (unsigned short)fa_bf16bits(p[base + 2 * k + 1]);
        pw[G][k] = (hi_b << 16) | lo;
      }
    }

    // ---- PV: O[q][d] += P[q][kv] @ V[kv][d] -----------------------------
#pragma unroll
    for (int c2 = 0; c2 < 4; ++c2) {  // kv chunks of 16
      unsigned int paw[4];
#pragma unroll
      for (int k = 0; k < 2; ++k) {
        auto pr = __builtin_amdgcn_permlane32_swap(
            (int)pw[2 * c2][k], (int)pw[2 * c2 + 1][k], false, false);
        paw[k] = (unsigned int)pr[0];
        paw[k + 2] = (unsigned int)pr[1];
      }
      bf16x8 pa;
#pragma unroll
      for (int k = 0; k < 4; ++k) {
        pa[2 * k] = (short)(paw[k] & 0xFFFF);
        pa[2 * k + 1] = (short)(paw[k] >> 16);
      }
#pragma unroll
      for (int t = 0; t < NTO; ++t) {   // d cols, 32 each
        const int dcol = t * 32 + ln;
        bf16x8 vb = *reinterpret_cast<const bf16x8*>(
            &vt_lds[cur][dcol][(c2 * 16 + hi * 8) ^ (dcol & 0x38)]);
        o_acc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa, vb, o_acc[t],
                                                           0, 0, 0);
      }
    }

    // stage tile kt+1 into the spare buffer (loop-top barrier publishes)
    if (kt + 1 < n_kv_tiles) {
      const int nxt = cur ^ 1;
#pragma unroll
      for (int sweep = 0; sweep < SWEEPS; ++sweep) {
        const int i = tid * 8 + sweep * (FA_WAVES * 64 * 8);
        if (i < ELEMS) {
          const int kr = i / D;
          const int kc = i % D;
          *reinterpret_cast<bf16x8*>(
              k_raw[nxt] + kr * SWB + kswz(kr, kc * 2)) = k_reg[sweep];
          bf16x8 vv = v_reg[sweep];
#pragma unroll
          for (int j = 0; j < 8; ++j)
            vt_lds[nxt][kc + j][kr ^ ((kc + j) & 0x38)] = vv[j];
        }
      }
      cur = nxt;
    }
  }

  // ---- epilogue ---------------------------------------------------------
  if (!q_active) return;
  const float my_inv_l = (l_run > 0.f) ? 1.f / l_run : 0.f;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int qrow = q0w + (r & 3) + 8 * (r >> 2) + 4 * hi;
    const float inv_l =
        __shfl(my_inv_l, (r & 3) + 8 * (r >> 2) + 4 * hi, 64);
#pragma unroll
    for (int t = 0; t < NTO; ++t) {
      Op[(long)qrow * D + t * 32 + ln] =
          __float2bfloat16(o_acc[t][r] * inv_l);
    }
  }
  if (hi == 0 && LSE) {
    LSE[bh * s + q0w + ln] = m_run + logf(fmaxf(l_run, 1e-30f));
  }
}

extern "C" void fs_flash_attn_fwd_v3(const void* q, const void* k,
                                     const void* v, void* o, float* lse,
                                     const int* klens, int b, int h, int s,
                                     int d, int causal, float scale,
                                     float drop_p, unsigned long long seed,
                                     hipStream_t stream) {
  dim3 grid((s + V3_QBLK - 1) / V3_QBLK, h, b);
  dim3 block(FA_WAVES * 64);
  const unsigned int dth =
      (drop_p > 0.f) ? (unsigned int)(drop_p * 4294967296.0) : 0u;
  const float ksc = (drop_p > 0.f) ? 1.f / (1.f - drop_p) : 1.f;
#define V3F(DD, CC, DR) hipLaunchKernelGGL( \
    (flash_attn_fwd_v3_kernel<DD, CC, DR>), \
    grid, block, 0, stream, (const bf16_t*)q, (const bf16_t*)k, \
    (const bf16_t*)v, (bf16_t*)o, lse, klens, b, h, s, scale, dth, ksc, seed)
  if (dth) {
    if (d == 128 && causal) V3F(128, true, true);
    else if (d == 128) V3F(128, false, true);
    else if (d == 96 && causal) V3F(96, true, true);
    else if (d == 96) V3F(96, false, true);
    else if (d == 64 && causal) V3F(64, true, true);
    else V3F(64, false, true);
  } else {
    if (d == 128 && causal) V3F(128, true, false);
    else if (d == 128) V3F(128, false, false);
    else if (d == 96 && causal) V3F(96, true, false);
    else if (d == 96) V3F(96, false, false);
    else if (d == 64 && causal) V3F(64, true, false);
    else V3F(64, false, false);
  }
#undef V3F
}

// ===========================================================================
// BACKWARD v3 — swapped 32x32 schedule (causal, D=128)
// ===========================================================================
// Same design move as forward v3: orient every S-like MFMA so the softmax
// row variable (q for dq, q-column for dkv) is lane-local — lse/delta
// become per-lane scalar loads, masks are per-register compares, and the
// only cross-lane traffic is the compact P/dS LDS exchange.

#define B3_FB 32  // streamed tile width

// q-major: wave owns 32 q rows (block 256), streams KV in 32-tiles.
template <int D, bool CAUSAL, bool HAS_DROP>
__global__ __launch_bounds__(FA_WAVES * 64)
void flash_attn_bwd_dq_v3_kernel(const bf16_t* __restrict__ Q,
                                 const bf16_t* __restrict__ K,
                                 const bf16_t* __restrict__ V,
                                 const bf16_t* __restrict__ dO,
                                 const float* __restrict__ LSE,
                                 const float* __restrict__ Delta,
                                 bf16_t* __restrict__ dQ,
                                 const int* __restrict__ klens,
                                 int b, int h, int s, float scale,
                                 unsigned int drop_thresh, float keep_scale,
                                 unsigned long long drop_seed) {
  constexpr int NC = D / 16;
  constexpr int NTO = D / 32;
  constexpr int SWB = fa_swb(D);

  __shared__ char k_raw[B3_FB * SWB];                  // K rows, swizzled
  __shared__ char v_raw[B3_FB * SWB];                  // V rows, swizzled
  __shared__ short kt_lds[D][B3_FB + FA_VPAD];         // K^T (kv-swizzled)
  __shared__ short p_x[FA_WAVES][32][B3_FB + FA_VPAD];

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int hi = lane >> 5;
  const int ln = lane & 31;
  const long bh = (long)blockIdx.z * h + blockIdx.y;
  const bf16_t* Qp = Q + bh * s * D;
  const bf16_t* Kp = K + bh * s * D;
  const bf16_t* Vp = V + bh * s * D;
  const bf16_t* dOp = dO + bh * s * D;
  bf16_t* dQp = dQ + bh * s * D;
  const float* lse = LSE + bh * s;
  const float* dlt = Delta + bh * s;

  const int klen = CAUSAL ? s : min(klens ? klens[blockIdx.z] : s, s);
  const int q0w = blockIdx.x * V3_QBLK + wave * 32;
  const bool q_active = q0w < s;
  const int q0c = q_active ? q0w : s - 32;
  const int my_q = q0c + ln;

  // Q (pre-scaled) and dO rows in registers (B-fragments)
  bf16x8 q_frag[NC], do_frag[NC];
#pragma unroll
  for (int c = 0; c < NC; ++c) {
    bf16x8 raw = *reinterpret_cast<const bf16x8*>(
        Qp + (long)my_q * D + c * 16 + hi * 8);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      unsigned short u = (unsigned short)raw[j];
      raw[j] = fa_bf16bits(
          __uint_as_float(((unsigned int)u) << 16) * scale);
    }
    q_frag[c] = raw;
    do_frag[c] = *reinterpret_cast<const bf16x8*>(
        dOp + (long)my_q * D + c * 16 + hi * 8);
  }
  const float lse_r = lse[my_q];
  const float dlt_r = dlt[my_q];

  f32x16 dq_acc[NTO];
#pragma unroll
  for (int t = 0; t < NTO; ++t) {
#pragma unroll
    for (int r = 0; r < 16; ++r) dq_acc[t][r] = 0.f;
  }

  int n_kv_tiles;
  if (CAUSAL) {
    const int q_hi_row = min(blockIdx.x * V3_QBLK + V3_QBLK, s) - 1;
    n_kv_tiles = (q_hi_row / B3_FB) + 1;
  } else {
    n_kv_tiles = (klen + B3_FB - 1) / B3_FB;
  }

  for (int kt = 0; kt < n_kv_tiles; ++kt) {
    const int k_base = kt * B3_FB;
    __syncthreads();
    {
      const int tid = threadIdx.x;
      for (int i = tid * 8; i < B3_FB * D; i += FA_WAVES * 64 * 8) {
        const int kr = i / D;
        const int kc = i % D;
        const long krg = min(k_base + kr, s - 1);
        bf16x8 kk = *reinterpret_cast<const bf16x8*>(Kp + krg * D + kc);
        *reinterpret_cast<bf16x8*>(k_raw + kr * SWB + kswz(kr, kc * 2)) = kk;
        bf16x8 vv = *reinterpret_cast<const bf16x8*>(Vp + krg * D + kc);
        *reinterpret_cast<bf16x8*>(v_raw + kr * SWB + kswz(kr, kc * 2)) = vv;
#pragma unroll
        for (int j = 0; j < 8; ++j)
          kt_lds[kc + j][kr ^ ((kc + j) & 0x18)] = kk[j];
      }
    }
    __syncthreads();

    // S^T = K (sQ)^T and dP^T = V dO^T : one 32x32 kv-tile each
    f32x16 st, dpt;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      st[r] = 0.f;
      dpt[r] = 0.f;
    }
    {
      const int krow = ln;
#pragma unroll
      for (int c = 0; c < NC; ++c) {
        bf16x8 ka = *reinterpret_cast<const bf16x8*>(
            k_raw + krow * SWB + kswz(krow, (c * 16 + hi * 8) * 2));
        st = __builtin_amdgcn_mfma_f32_32x32x16_bf16(ka, q_frag[c], st,
                                                     0, 0, 0);
        bf16x8 va = *reinterpret_cast<const bf16x8*>(
            v_raw + krow * SWB + kswz(krow, (c * 16 + hi * 8) * 2));
        dpt = __builtin_amdgcn_mfma_f32_32x32x16_bf16(va, do_frag[c], dpt,
                                                      0, 0, 0);
      }
    }
    // lane holds, for its OWN q row: kv = k_base + crow(r,hi)

    // dS = P * (dP - delta) * scale, in-register; exchange by q row
#pragma unroll
    for (int g = 0; g < 4; ++g) {
      short pk[4] __attribute__((aligned(8)));
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int r = g * 4 + j;
        const int kcol = k_base + (r & 3) + 8 * (r >> 2) + 4 * hi;
        const bool masked = CAUSAL ? (kcol > my_q) : (kcol >= klen);
        float pv = masked ? 0.f : __expf(st[r] - lse_r);
        float dpv = dpt[r];
        if (HAS_DROP && drop_thresh) {
          const unsigned long long idx =
              (bh * (unsigned long long)s + my_q) *
                  (unsigned long long)s + kcol;
          dpv = (fa_hash(drop_seed, idx) < drop_thresh)
                    ? 0.f : dpv * keep_scale;
        }
        pk[j] = fa_bf16bits(pv * (dpv - dlt_r) * scale);
      }
      *reinterpret_cast<long*>(&p_x[wave][ln][8 * g + 4 * hi]) =
          *reinterpret_cast<const long*>(pk);
    }
    __builtin_amdgcn_s_waitcnt(0);

    // dQ += dS @ K : A = dS [32q x 16kv], B = K^T
#pragma unroll
    for (int c2 = 0; c2 < 2; ++c2) {
      bf16x8 pa = *reinterpret_cast<const bf16x8*>(
          &p_x[wave][ln][c2 * 16 + hi * 8]);
#pragma unroll
      for (int t = 0; t < NTO; ++t) {
        const int dcol = t * 32 + ln;
        bf16x8 kb = *reinterpret_cast<const bf16x8*>(
            &kt_lds[dcol][(c2 * 16 + hi * 8) ^ (dcol & 0x18)]);
        dq_acc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa, kb,
                                                            dq_acc[t],
                                                            0, 0, 0);
      }
    }
  }

  if (!q_active) return;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int qrow = q0w + (r & 3) + 8 * (r >> 2) + 4 * hi;
#pragma unroll
    for (int t = 0; t < NTO; ++t)
      dQp[(long)qrow * D + t * 32 + ln] = __float2bfloat16(dq_acc[t][r]);
  }
}

// KV-major: wave owns 32 kv rows (block 256), streams Q/dO in 32-tiles.
// VGPR budget forces a T16-style time-share: the wave's own K rows load
// into a register block for the St MFMAs, then V overwrites the same
// block for the dPt MFMAs (rows are L2-hot; the loads hide under MFMA).
// The softmax scale folds into the exp (st*scale - lse) so K stays raw.
template <int D, bool CAUSAL, bool HAS_DROP>
__global__ __launch_bounds__(FA_WAVES * 64)
void flash_attn_bwd_dkv_v3_kernel(const bf16_t* __restrict__ Q,
                                  const bf16_t* __restrict__ K,
                                  const bf16_t* __restrict__ V,
                                  const bf16_t* __restrict__ dO,
                                  const float* __restrict__ LSE,
                                  const float* __restrict__ Delta,
                                  bf16_t* __restrict__ dK,
                                  bf16_t* __restrict__ dV,
                                  const int* __restrict__ klens,
                                  int b, int h, int s, float scale,
                                  unsigned int drop_thresh, float keep_scale,
                                  unsigned long long drop_seed) {
  constexpr int NC = D / 16;
  constexpr int NTO = D / 32;
  constexpr int SWB = fa_swb(D);

  __shared__ char q_raw[B3_FB * SWB];                   // Q rows, swizzled
  __shared__ char do_raw[B3_FB * SWB];                  // dO rows, swizzled
  __shared__ short qt_lds[D][B3_FB + FA_VPAD];          // Q^T
  __shared__ short dot_lds[D][B3_FB + FA_VPAD];         // dO^T
  __shared__ short p_x[FA_WAVES][32][B3_FB + FA_VPAD];   // Pt by kv row
  __shared__ short ds_x[FA_WAVES][32][B3_FB + FA_VPAD];  // dSt by kv row

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int hi = lane >> 5;
  const int ln = lane & 31;
  const long bh = (long)blockIdx.z * h + blockIdx.y;
  const bf16_t* Qp = Q + bh * s * D;
  const bf16_t* Kp = K + bh * s * D;
  const bf16_t* Vp = V + bh * s * D;
  const bf16_t* dOp = dO + bh * s * D;
  bf16_t* dKp = dK + bh * s * D;
  bf16_t* dVp = dV + bh * s * D;
  const float* lse = LSE + bh * s;
  const float* dlt = Delta + bh * s;

  const int klen = CAUSAL ? s : min(klens ? klens[blockIdx.z] : s, s);
  const int kv0_blk = blockIdx.x * V3_QBLK;
  const int kv0w = kv0_blk + wave * 32;
  const bool kv_active = kv0w < s;
  const int kv0c = kv_active ? kv0w : s - 32;
  const int my_kv = kv0c + ln;

  // dv/dk in the 32x32 layout: NTO d-tiles x 16 f32 each
  f32x16 dv_acc[NTO], dk_acc[NTO];
#pragma unroll
  for (int t = 0; t < NTO; ++t) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      dv_acc[t][r] = 0.f;
      dk_acc[t][r] = 0.f;
    }
  }

  const int first_qt = CAUSAL ? kv0_blk / B3_FB : 0;
  const int n_q_tiles = s / B3_FB;

  for (int qt = first_qt; qt < n_q_tiles; ++qt) {
    const int q_base = qt * B3_FB;
    __syncthreads();
    {
      const int tid = threadIdx.x;
      for (int i = tid * 8; i < B3_FB * D; i += FA_WAVES * 64 * 8) {
        const int qr = i / D;
        const int qc = i % D;
        bf16x8 qq = *reinterpret_cast<const bf16x8*>(
            Qp + (long)(q_base + qr) * D + qc);
        *reinterpret_cast<bf16x8*>(q_raw + qr * SWB + kswz(qr, qc * 2)) = qq;
        bf16x8 dd = *reinterpret_cast<const bf16x8*>(
            dOp + (long)(q_base + qr) * D + qc);
        *reinterpret_cast<bf16x8*>(do_raw + qr * SWB + kswz(qr, qc * 2)) =
            dd;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          qt_lds[qc + j][qr ^ ((qc + j) & 0x18)] = qq[j];
          dot_lds[qc + j][qr ^ ((qc + j) & 0x18)] = dd[j];
        }
      }
    }
    __syncthreads();

    // St = K Q^T (raw K; scale folded into the exp below)
    f32x16 st;
#pragma unroll
    for (int r = 0; r < 16; ++r) st[r] = 0.f;
#pragma unroll
    for (int c = 0; c < NC; ++c) {
      bf16x8 ka = *reinterpret_cast<const bf16x8*>(
          Kp + (long)my_kv * D + c * 16 + hi * 8);
      bf16x8 qb = *reinterpret_cast<const bf16x8*>(
          q_raw + ln * SWB + kswz(ln, (c * 16 + hi * 8) * 2));
      st = __builtin_amdgcn_mfma_f32_32x32x16_bf16(ka, qb, st, 0, 0, 0);
    }
    // dPt = V dO^T (same register pattern, V overwrites K's slots)
    f32x16 dpt;
#pragma unroll
    for (int r = 0; r < 16; ++r) dpt[r] = 0.f;
#pragma unroll
    for (int c = 0; c < NC; ++c) {
      bf16x8 va = *reinterpret_cast<const bf16x8*>(
          Vp + (long)my_kv * D + c * 16 + hi * 8);
      bf16x8 db = *reinterpret_cast<const bf16x8*>(
          do_raw + ln * SWB + kswz(ln, (c * 16 + hi * 8) * 2));
      dpt = __builtin_amdgcn_mfma_f32_32x32x16_bf16(va, db, dpt, 0, 0, 0);
    }

    const int qcol = q_base + ln;
    const float lse_c = lse[qcol];
    const float dlt_c = dlt[qcol];

    // Pt AND dSt by kv row, one exchange pass: st/dpt registers die
    // here (keeping them through the dV MFMAs spilled 128 B/lane)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int kvrow_loc = (r & 3) + 8 * (r >> 2) + 4 * hi;
      const int kvrow = kv0c + kvrow_loc;
      const bool masked = CAUSAL ? (qcol < kvrow) : (kvrow >= klen);
      float pv = masked ? 0.f : __expf(st[r] * scale - lse_c);
      float dm = 1.f;
      if (HAS_DROP && drop_thresh) {
        const unsigned long long idx =
            (bh * (unsigned long long)s + qcol) *
                (unsigned long long)s + kvrow;
        dm = (fa_hash(drop_seed, idx) < drop_thresh) ? 0.f : keep_scale;
      }
      p_x[wave][kvrow_loc][ln] = fa_bf16bits(pv * dm);  // dropped P -> dV
      ds_x[wave][kvrow_loc][ln] =
          fa_bf16bits(pv * (dm * dpt[r] - dlt_c) * scale);  // dSt -> dK
    }
    __builtin_amdgcn_s_waitcnt(0);

    // dV += Pt dO : A = Pt [32kv x 16q-chunk] own-row, B = dO^T
#pragma unroll
    for (int c2 = 0; c2 < 2; ++c2) {
      bf16x8 pa = *reinterpret_cast<const bf16x8*>(
          &p_x[wave][ln][c2 * 16 + hi * 8]);
#pragma unroll
      for (int t = 0; t < NTO; ++t) {
        const int dcol = t * 32 + ln;
        bf16x8 bfrag = *reinterpret_cast<const bf16x8*>(
            &dot_lds[dcol][(c2 * 16 + hi * 8) ^ (dcol & 0x18)]);
        dv_acc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            pa, bfrag, dv_acc[t], 0, 0, 0);
      }
    }

    // dK += dSt Q (dSt already exchanged above)
#pragma unroll
    for (int c2 = 0; c2 < 2; ++c2) {
      bf16x8 pa = *reinterpret_cast<const bf16x8*>(
          &ds_x[wave][ln][c2 * 16 + hi * 8]);
#pragma unroll
      for (int t = 0; t < NTO; ++t) {
        const int dcol = t * 32 + ln;
        bf16x8 bfrag = *reinterpret_cast<const bf16x8*>(
            &qt_lds[dcol][(c2 * 16 + hi * 8) ^ (dcol & 0x18)]);
        dk_acc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            pa, bfrag, dk_acc[t], 0, 0, 0);
      }
    }
  }

  if (!kv_active) return;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int kvrow = kv0w + (r & 3) + 8 * (r >> 2) + 4 * hi;
#pragma unroll
    for (int t = 0; t < NTO; ++t) {
      dKp[(long)kvrow * D + t * 32 + ln] = __float2bfloat16(dk_acc[t][r]);
      dVp[(long)kvrow * D + t * 32 + ln] = __float2bfloat16(dv_acc[t][r]);
    }
  }
}

template <int D, bool CAUSAL, bool HAS_DROP>
static void launch_bwd_v3(const void* q, const void* k, const void* v,
                          const void* o, const void* dout, const float* lse,
                          void* dq, void* dk, void* dv, float* delta_ws,
                          const int* klens, int b, int h, int s, float scale,
                          unsigned int dth, float ksc,
                          unsigned long long seed, hipStream_t stream) {
  const long rows = (long)b * h * s;
  {
    long blocks = (rows + 3) / 4;
    if (blocks > 4096) blocks = 4096;
    hipLaunchKernelGGL((flash_delta_kernel<D>), dim3((unsigned)blocks),
                       dim3(256), 0, stream, (const bf16_t*)dout,
                       (const bf16_t*)o, delta_ws, rows);
  }
  dim3 grid((s + V3_QBLK - 1) / V3_QBLK, h, b);
  dim3 block(FA_WAVES * 64);
  hipLaunchKernelGGL((flash_attn_bwd_dq_v3_kernel<D, CAUSAL, HAS_DROP>),
                     grid, block, 0, stream, (const bf16_t*)q,
                     (const bf16_t*)k, (const bf16_t*)v, (const bf16_t*)dout,
                     lse, delta_ws, (bf16_t*)dq, klens, b, h, s, scale, dth,
                     ksc, seed);
  hipLaunchKernelGGL((flash_attn_bwd_dkv_v3_kernel<D, CAUSAL, HAS_DROP>),
                     grid, block, 0, stream, (const bf16_t*)q,
                     (const bf16_t*)k, (const bf16_t*)v, (const bf16_t*)dout,
                     lse, delta_ws, (bf16_t*)dk, (bf16_t*)dv, klens, b, h, s,
                     scale, dth, ksc, seed);
}

extern "C" void fs_flash_attn_bwd_v3(const void* q, const void* k,
                                     const void* v, const void* o,
                                     const void* dout, const float* lse,
                                     void* dq, void* dk, void* dv,
                                     float* delta_ws, const int* klens,
                                     int b, int h, int s, int d, int causal,
                                     float scale, float drop_p,
                                     unsigned long long seed,
                                     hipStream_t stream) {
  const unsigned int dth =
      (drop_p > 0.f) ? (unsigned int)(drop_p * 4294967296.0) : 0u;
  const float ksc = (drop_p > 0.f) ? 1.f / (1.f - drop_p) : 1.f;
#define BWD3(DD, CC, DR) launch_bwd_v3<DD, CC, DR>( \
      q, k, v, o, dout, lse, dq, dk, dv, delta_ws, klens, b, h, s, scale, \
      dth, ksc, seed, stream)
  if (dth) {
    if (d == 128 && causal) BWD3(128, true, true);
    else if (d == 128) BWD3(128, false, true);
    else if (d == 96 && causal) BWD3(96, true, true);
    else if (d == 96) BWD3(96, false, true);
    else if (d == 64 && causal) BWD3(64, true, true);
    else BWD3(64, false, true);
  } else {
    if (d == 128 && causal) BWD3(128, true, false);
    else if (d == 128) BWD3(128, false, false);
    else if (d == 96 && causal) BWD3(96, true, false);
    else if (d == 96) BWD3(96, false, false);
    else if (d == 64 && causal) BWD3(64, true, false);
    else BWD3(64, false, false);
  }
#undef BWD3
}
