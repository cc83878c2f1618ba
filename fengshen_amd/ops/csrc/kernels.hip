// fengshen_amd fused kernels for MI355X (gfx950, CDNA4).
//
// Design (per /opt/skills/guides/cdna_hip_programming.md):
//   * all ops here are memory-bound: vectorized bf16 loads (short8, 16B/lane),
//     one global read + one global write per tensor where possible
//   * rows staged through LDS (160 KiB/CU) so softmax/norms read HBM once
//   * wave64 shuffle reductions, block reduction via LDS
//   * grid-stride loops, 256-thread blocks
//
// Replaces (behavioral parity, redesigned for wave64 — NOT translated):
//   scaled_masked_softmax_cuda + scaled_upper_triang_masked_softmax_cuda
//     (reference fused_kernels/*.h warp-softmax, warp=32, sk<=2048 cap;
//      ours: LDS row softmax, any sk up to 40960)
//   apex fused layer norm (reference layer_norm_cuda.cpp, vestigial)
//   RMSNorm/rotary/SwiGLU/bias-GELU (reference runs these eager/jit)
//   DeepSpeed FusedAdam (multi-bucket flat fused AdamW)

#include "common.h"

enum FsDtype { FS_F32 = 0, FS_BF16 = 1, FS_F16 = 2 };

// ===========================================================================
// RMSNorm
// ===========================================================================
template <typename T>
__global__ void rms_norm_fwd_kernel(const T* __restrict__ x,
                                    const T* __restrict__ w,
                                    T* __restrict__ out,
                                    float* __restrict__ invrms,
                                    int rows, int H, float eps) {
  extern __shared__ float lds[];          // [H] row staging
  __shared__ float red[32];
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* xr = x + (long)row * H;
    T* yr = out + (long)row * H;
    float ss = 0.f;
    for (int c = threadIdx.x * 8; c < H; c += blockDim.x * 8) {
      float v[8];
      load8<T>(xr + c, v);
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        lds[c + i] = v[i];
        ss += v[i] * v[i];
      }
    }
    ss = block_reduce_sum(ss, red);
    float ir = rsqrtf(ss / H + eps);
    if (threadIdx.x == 0 && invrms) invrms[row] = ir;
    for (int c = threadIdx.x * 8; c < H; c += blockDim.x * 8) {
      float wv[8], o[8];
      load8<T>(w + c, wv);
#pragma unroll
      for (int i = 0; i < 8; ++i) o[i] = lds[c + i] * ir * wv[i];
      store8<T>(yr + c, o);
    }
    __syncthreads();
  }
}

// bwd: gx = ir*(gy*w - xhat * mean(gy*w*xhat)); gw(+)= gy*xhat (fp32 atomics)
template <typename T>
__global__ void rms_norm_bwd_kernel(const T* __restrict__ gy,
                                    const T* __restrict__ x,
                                    const T* __restrict__ w,
                                    const float* __restrict__ invrms,
                                    T* __restrict__ gx,
                                    float* __restrict__ gw_f32,
                                    int rows, int H) {
  extern __shared__ float lds[];  // [3*H]: xhat row, gyw row, gw accumulator
  __shared__ float red[32];
  float* xhat_s = lds;
  float* gyw_s = lds + H;
  float* gw_s = lds + 2 * H;
  for (int c = threadIdx.x; c < H; c += blockDim.x) gw_s[c] = 0.f;
  __syncthreads();
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* xr = x + (long)row * H;
    const T* gr = gy + (long)row * H;
    T* gxr = gx + (long)row * H;
    const float ir = invrms[row];
    float dot = 0.f;
    for (int c = threadIdx.x * 8; c < H; c += blockDim.x * 8) {
      float xv[8], gv[8], wv[8];
      load8<T>(xr + c, xv);
      load8<T>(gr + c, gv);
      load8<T>(w + c, wv);
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        float xh = xv[i] * ir;
        float gw_ = gv[i] * wv[i];
        xhat_s[c + i] = xh;
        gyw_s[c + i] = gw_;
        dot += gw_ * xh;
        gw_s[c + i] += gv[i] * xh;  // block-local; flushed once at the end
      }
    }
    dot = block_reduce_sum(dot, red) / H;
    for (int c = threadIdx.x * 8; c < H; c += blockDim.x * 8) {
      float o[8];
#pragma unroll
      for (int i = 0; i < 8; ++i)
        o[i] = ir * (gyw_s[c + i] - xhat_s[c + i] * dot);
      store8<T>(gxr + c, o);
    }
    __syncthreads();
  }
  for (int c = threadIdx.x; c < H; c += blockDim.x)
    atomicAdd(&gw_f32[c], gw_s[c]);
}

// ===========================================================================
// LayerNorm
// ===========================================================================
template <typename T>
__global__ void layer_norm_fwd_kernel(const T* __restrict__ x,
                                      const T* __restrict__ w,
                                      const T* __restrict__ b,
                                      T* __restrict__ out,
                                      float* __restrict__ mean_out,
                                      float* __restrict__ invstd_out,
                                      int rows, int H, float eps) {
  extern __shared__ float lds[];  // [H]
  __shared__ float red[32];
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* xr = x + (long)row * H;
    T* yr = out + (long)row * H;
    float s = 0.f;
    for (int c = threadIdx.x * 8; c < H; c += blockDim.x * 8) {
      float v[8];
      load8<T>(xr + c, v);
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        lds[c + i] = v[i];
        s += v[i];
      }
    }
    float mu = block_reduce_sum(s, red) / H;
    float var = 0.f;
    for (int c = threadIdx.x * 8; c < H; c += blockDim.x * 8) {
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        float d = lds[c + i] - mu;
        var += d * d;
      }
    }
    var = block_reduce_sum(var, red) / H;
    float is = rsqrtf(var + eps);
    if (threadIdx.x == 0) {
      if (mean_out) mean_out[row] = mu;
      if (invstd_out) invstd_out[row] = is;
    }
    for (int c = threadIdx.x * 8; c < H; c += blockDim.x * 8) {
      float wv[8], bv[8], o[8];
      load8<T>(w + c, wv);
      if (b) load8<T>(b + c, bv);
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        o[i] = (lds[c + i] - mu) * is * wv[i] + (b ? bv[i] : 0.f);
      }
      store8<T>(yr + c, o);
    }
    __syncthreads();
  }
}

template <typename T>
__global__ void layer_norm_bwd_kernel(const T* __restrict__ gy,
                                      const T* __restrict__ x,
                                      const T* __restrict__ w,
                                      const float* __restrict__ mean,
                                      const float* __restrict__ invstd,
                                      T* __restrict__ gx,
                                      float* __restrict__ gw_f32,
                                      float* __restrict__ gb_f32,
                                      int rows, int H) {
  extern __shared__ float lds[];  // [4*H]: xhat, gyw, gw acc, gb acc
  __shared__ float red[32];
  float* xhat_s = lds;
  float* gyw_s = lds + H;
  float* gw_s = lds + 2 * H;
  float* gb_s = lds + 3 * H;
  for (int c = threadIdx.x; c < H; c += blockDim.x) {
    gw_s[c] = 0.f;
    gb_s[c] = 0.f;
  }
  __syncthreads();
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* xr = x + (long)row * H;
    const T* gr = gy + (long)row * H;
    T* gxr = gx + (long)row * H;
    const float mu = mean[row];
    const float is = invstd[row];
    float dot = 0.f, gsum = 0.f;
    for (int c = threadIdx.x * 8; c < H; c += blockDim.x * 8) {
      float xv[8], gv[8], wv[8];
      load8<T>(xr + c, xv);
      load8<T>(gr + c, gv);
      load8<T>(w + c, wv);
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        float xh = (xv[i] - mu) * is;
        float gw_ = gv[i] * wv[i];
        xhat_s[c + i] = xh;
        gyw_s[c + i] = gw_;
        dot += gw_ * xh;
        gsum += gw_;
        gw_s[c + i] += gv[i] * xh;
        gb_s[c + i] += gv[i];
      }
    }
    dot = block_reduce_sum(dot, red) / H;
    gsum = block_reduce_sum(gsum, red) / H;
    for (int c = threadIdx.x * 8; c < H; c += blockDim.x * 8) {
      float o[8];
#pragma unroll
      for (int i = 0; i < 8; ++i)
        o[i] = is * (gyw_s[c + i] - gsum - xhat_s[c + i] * dot);
      store8<T>(gxr + c, o);
    }
    __syncthreads();
  }
  for (int c = threadIdx.x; c < H; c += blockDim.x) {
    atomicAdd(&gw_f32[c], gw_s[c]);
    if (gb_f32) atomicAdd(&gb_f32[c], gb_s[c]);
  }
}

// ===========================================================================
// Softmax family (fused scale + mask + softmax, fp32 accumulation in LDS)
// ===========================================================================
// mode 0: plain scaled softmax
// mode 1: padding mask (uint8, 1 = masked), mask row layout [mb,1,sq,sk]
// mode 2: causal (upper-triangular masked by index compare, no mask tensor)
template <typename T, int MODE>
__global__ void scaled_softmax_fwd_kernel(const T* __restrict__ x,
                                          const unsigned char* __restrict__ mask,
                                          T* __restrict__ out, float scale,
                                          int rows, int sk, int sq,
                                          int np, int mask_batches,
                                          int mask_sq) {
  extern __shared__ float lds[];  // [sk]
  __shared__ float red[32];
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* xr = x + (long)row * sk;
    T* yr = out + (long)row * sk;
    const int si = row % sq;                 // query position
    const unsigned char* mr = nullptr;
    if (MODE == 1) {
      const int bi = row / (np * sq);
      const int mb = (mask_batches == 1) ? 0 : bi;
      const int msi = (mask_sq == 1) ? 0 : si;  // broadcast [b,1,1,sk] masks
      mr = mask + ((long)mb * mask_sq + msi) * sk;
    }
    const int limit = (MODE == 2) ? (si + 1) : sk;

    float mx = -INFINITY;
    int c = threadIdx.x * 8;
    for (; c + 7 < limit; c += blockDim.x * 8) {
      float v[8];
      load8<T>(xr + c, v);
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        float val = v[i] * scale;
        if (MODE == 1 && mr[c + i]) val = -10000.f;
        lds[c + i] = val;
        mx = fmaxf(mx, val);
      }
    }
    // scalar tail: elements in [limit & ~7, limit)
    for (int t = (limit & ~7) + threadIdx.x; t < limit; t += blockDim.x) {
      float val = to_f32<T>(xr[t]) * scale;
      if (MODE == 1 && mr[t]) val = -10000.f;
      lds[t] = val;
      mx = fmaxf(mx, val);
    }
    mx = block_reduce_max(mx, red);
    float sum = 0.f;
    for (int t = threadIdx.x; t < limit; t += blockDim.x) {
      float e = __expf(lds[t] - mx);
      lds[t] = e;
      sum += e;
    }
    sum = block_reduce_sum(sum, red);
    const float inv = 1.f / sum;
    for (int t = threadIdx.x; t < limit; t += blockDim.x)
      lds[t] *= inv;
    __syncthreads();
    for (c = threadIdx.x * 8; c + 7 < sk; c += blockDim.x * 8) {
      float o[8];
#pragma unroll
      for (int i = 0; i < 8; ++i) o[i] = (c + i < limit) ? lds[c + i] : 0.f;
      store8<T>(yr + c, o);
    }
    for (int t = (sk & ~7) + threadIdx.x; t < sk; t += blockDim.x)
      yr[t] = from_f32<T>(t < limit ? lds[t] : 0.f);
    __syncthreads();
  }
}

// bwd (shared by all modes): dx = scale * y * (gy - sum(gy*y))
template <typename T>
__global__ void scaled_softmax_bwd_kernel(const T* __restrict__ gy,
                                          const T* __restrict__ y,
                                          T* __restrict__ gx, float scale,
                                          int rows, int sk) {
  extern __shared__ float lds[];  // [2*sk]: y, gy
  __shared__ float red[32];
  float* ys = lds;
  float* gs = lds + sk;
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* yr = y + (long)row * sk;
    const T* gr = gy + (long)row * sk;
    T* gxr = gx + (long)row * sk;
    float dot = 0.f;
    int c = threadIdx.x * 8;
    for (; c + 7 < sk; c += blockDim.x * 8) {
      float yv[8], gv[8];
      load8<T>(yr + c, yv);
      load8<T>(gr + c, gv);
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        ys[c + i] = yv[i];
        gs[c + i] = gv[i];
        dot += yv[i] * gv[i];
      }
    }
    for (int t = (sk & ~7) + threadIdx.x; t < sk; t += blockDim.x) {
      float yv = to_f32<T>(yr[t]);
      float gv = to_f32<T>(gr[t]);
      ys[t] = yv;
      gs[t] = gv;
      dot += yv * gv;
    }
    dot = block_reduce_sum(dot, red);
    for (c = threadIdx.x * 8; c + 7 < sk; c += blockDim.x * 8) {
      float o[8];
#pragma unroll
      for (int i = 0; i < 8; ++i)
        o[i] = scale * ys[c + i] * (gs[c + i] - dot);
      store8<T>(gxr + c, o);
    }
    for (int t = (sk & ~7) + threadIdx.x; t < sk; t += blockDim.x)
      gxr[t] = from_f32<T>(scale * ys[t] * (gs[t] - dot));
    __syncthreads();
  }
}

// ===========================================================================
// RoPE (half-rotation layout: pair (d, d+hn/2), shared cos/sin per pair)
// ===========================================================================
template <typename T, bool BWD>
__global__ void rope_kernel(const T* __restrict__ x, T* __restrict__ out,
                            const float* __restrict__ cos_t,
                            const float* __restrict__ sin_t,
                            long total_pairs, int s, int h2, int offset) {
  // x: [b*np, s, hn]; pairs indexed by (bn, si, d) with d in [0, h2)
  const long stride = (long)gridDim.x * blockDim.x * 8;
  for (long i = (long)(blockIdx.x * blockDim.x + threadIdx.x) * 8;
       i < total_pairs; i += stride) {
    // decompose on the first element of the 8-vector; the whole vector is
    // within one (bn, si) row because h2 % 8 == 0
    const long d = i % h2;
    const long si = (i / h2) % s;
    const long bn = i / ((long)h2 * s);
    const long base = (bn * s + si) * (2 * h2);
    const float* cr = cos_t + (si + offset) * (2 * h2) + d;
    const float* sr = sin_t + (si + offset) * (2 * h2) + d;
    float x1[8], x2[8], o1[8], o2[8];
    load8<T>(x + base + d, x1);
    load8<T>(x + base + d + h2, x2);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float c = cr[j];
      const float sn = sr[j];
      if (!BWD) {
        o1[j] = x1[j] * c - x2[j] * sn;
        o2[j] = x2[j] * c + x1[j] * sn;
      } else {  // transpose rotation
        o1[j] = x1[j] * c + x2[j] * sn;
        o2[j] = x2[j] * c - x1[j] * sn;
      }
    }
    store8<T>(out + base + d, o1);
    store8<T>(out + base + d + h2, o2);
  }
}

// ===========================================================================
// SwiGLU: packed [rows, 2h] -> out [rows, h]
// ===========================================================================
template <typename T>
__global__ void swiglu_fwd_kernel(const T* __restrict__ packed,
                                  T* __restrict__ out, long rows, int h) {
  const long total = rows * (long)h;
  const long stride = (long)gridDim.x * blockDim.x * 8;
  for (long i = (long)(blockIdx.x * blockDim.x + threadIdx.x) * 8; i < total;
       i += stride) {
    const long row = i / h;
    const long col = i % h;
    const T* g = packed + row * 2 * h + col;
    const T* u = g + h;
    float gv[8], uv[8], o[8];
    load8<T>(g, gv);
    load8<T>(u, uv);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float sig = 1.f / (1.f + __expf(-gv[j]));
      o[j] = gv[j] * sig * uv[j];
    }
    store8<T>(out + i, o);
  }
}

template <typename T>
__global__ void swiglu_bwd_kernel(const T* __restrict__ gy,
                                  const T* __restrict__ packed,
                                  T* __restrict__ gpacked, long rows, int h) {
  const long total = rows * (long)h;
  const long stride = (long)gridDim.x * blockDim.x * 8;
  for (long i = (long)(blockIdx.x * blockDim.x + threadIdx.x) * 8; i < total;
       i += stride) {
    const long row = i / h;
    const long col = i % h;
    const T* g = packed + row * 2 * h + col;
    const T* u = g + h;
    float gv[8], uv[8], go[8], dg[8], du[8];
    load8<T>(g, gv);
    load8<T>(u, uv);
    load8<T>(gy + i, go);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float sig = 1.f / (1.f + __expf(-gv[j]));
      const float silu = gv[j] * sig;
      dg[j] = go[j] * uv[j] * sig * (1.f + gv[j] * (1.f - sig));
      du[j] = go[j] * silu;
    }
    store8<T>(gpacked + row * 2 * h + col, dg);
    store8<T>(gpacked + row * 2 * h + col + h, du);
  }
}

// ===========================================================================
// bias-GELU (tanh approx, matches reference activations.py:60-77)
// ===========================================================================
__device__ __forceinline__ float gelu_tanh(float y) {
  return y * 0.5f * (1.f + tanhf(0.79788456f * y * (1.f + 0.044715f * y * y)));
}
__device__ __forceinline__ float gelu_tanh_grad(float y) {
  const float t = tanhf(0.79788456f * y * (1.f + 0.044715f * y * y));
  return 0.5f * y * ((1.f - t * t) * (0.79788456f + 0.1070322243f * y * y)) +
         0.5f * (1.f + t);
}

template <typename T, bool BWD>
__global__ void bias_gelu_kernel(const T* __restrict__ x,
                                 const T* __restrict__ bias,
                                 const T* __restrict__ gy,
                                 T* __restrict__ out, long rows, int h) {
  const long total = rows * (long)h;
  const long stride = (long)gridDim.x * blockDim.x * 8;
  for (long i = (long)(blockIdx.x * blockDim.x + threadIdx.x) * 8; i < total;
       i += stride) {
    const long col = i % h;
    float xv[8], bv[8], o[8];
    load8<T>(x + i, xv);
    load8<T>(bias + col, bv);
    if (!BWD) {
#pragma unroll
      for (int j = 0; j < 8; ++j) o[j] = gelu_tanh(xv[j] + bv[j]);
    } else {
      float gv[8];
      load8<T>(gy + i, gv);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        o[j] = gelu_tanh_grad(xv[j] + bv[j]) * gv[j];
    }
    store8<T>(out + i, o);
  }
}

// ===========================================================================
// fused AdamW over flat fp32 master (+ low-precision param copy-out)
// ===========================================================================
template <typename GT, typename OT>
__global__ void fused_adamw_kernel(float* __restrict__ master,
                                   const GT* __restrict__ grad,
                                   float* __restrict__ m,
                                   float* __restrict__ v,
                                   OT* __restrict__ out_param,
                                   long n, float lr, float beta1, float beta2,
                                   float eps, float wd, float bc1, float bc2) {
  // vectorized: 4 fp32 = 16B per lane for master/m/v (ZeRO shards are
  // 128-element aligned so n % 4 == 0 in practice; tail handled scalar)
  const long n4 = n & ~3L;
  const long stride = (long)gridDim.x * blockDim.x * 4;
  for (long i = (long)(blockIdx.x * blockDim.x + threadIdx.x) * 4; i < n4;
       i += stride) {
    float4_t p = *reinterpret_cast<float4_t*>(master + i);
    float4_t mj = *reinterpret_cast<float4_t*>(m + i);
    float4_t vj = *reinterpret_cast<float4_t*>(v + i);
    float g[4];
#pragma unroll
    for (int j = 0; j < 4; ++j) g[j] = to_f32<GT>(grad[i + j]);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float pj = p[j] * (1.f - lr * wd);
      float mm = beta1 * mj[j] + (1.f - beta1) * g[j];
      float vv = beta2 * vj[j] + (1.f - beta2) * g[j] * g[j];
      const float denom = sqrtf(vv / bc2) + eps;
      pj -= lr / bc1 * (mm / denom);
      p[j] = pj;
      mj[j] = mm;
      vj[j] = vv;
    }
    *reinterpret_cast<float4_t*>(master + i) = p;
    *reinterpret_cast<float4_t*>(m + i) = mj;
    *reinterpret_cast<float4_t*>(v + i) = vj;
    if (out_param) {
#pragma unroll
      for (int j = 0; j < 4; ++j) out_param[i + j] = from_f32<OT>(p[j]);
    }
  }
  // scalar tail
  for (long t = n4 + blockIdx.x * blockDim.x + threadIdx.x; t < n;
       t += (long)gridDim.x * blockDim.x) {
    float g = to_f32<GT>(grad[t]);
    float pj = master[t] * (1.f - lr * wd);
    float mm = m[t] = beta1 * m[t] + (1.f - beta1) * g;
    float vv = v[t] = beta2 * v[t] + (1.f - beta2) * g * g;
    pj -= lr / bc1 * (mm / (sqrtf(vv / bc2) + eps));
    master[t] = pj;
    if (out_param) out_param[t] = from_f32<OT>(pj);
  }
}

// ===========================================================================
// Wave-per-row softmax (sk <= 2048, sk % 8 == 0): the whole row lives in
// registers (<=32 fp32/lane), reductions are wave shuffles — no LDS, no
// __syncthreads.  4 rows per 256-thread block.  The LDS block-per-row
// kernel above remains the fallback for larger sk.
// ===========================================================================
template <typename T, int MODE>
__global__ void scaled_softmax_fwd_wave_kernel(
    const T* __restrict__ x, const unsigned char* __restrict__ mask,
    T* __restrict__ out, float scale, long rows, int sk, int sq,
    int np, int mask_batches, int mask_sq) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int nchunks = (sk + 511) / 512;
  float v[32];
  for (long row = (long)blockIdx.x * 4 + wid; row < rows;
       row += (long)gridDim.x * 4) {
    const T* xr = x + row * sk;
    T* yr = out + row * sk;
    const int si = (int)(row % sq);
    const unsigned char* mr = nullptr;
    if (MODE == 1) {
      const long bi = row / ((long)np * sq);
      const long mb = (mask_batches == 1) ? 0 : bi;
      const int msi = (mask_sq == 1) ? 0 : si;
      mr = mask + ((long)mb * mask_sq + msi) * sk;
    }
    const int limit = (MODE == 2) ? (si + 1) : sk;

    float mx = -INFINITY;
#pragma unroll 4
    for (int c = 0; c < nchunks; ++c) {
      const int base = c * 512 + lane * 8;
      if (base < sk) {
        float t[8];
        load8<T>(xr + base, t);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float val = t[j] * scale;
          if (MODE == 1 && mr[base + j]) val = -10000.f;
          if (MODE == 2 && base + j >= limit) val = -INFINITY;
          v[c * 8 + j] = val;
          mx = fmaxf(mx, val);
        }
      }
    }
    mx = wave_reduce_max(mx);
    float sum = 0.f;
#pragma unroll 4
    for (int c = 0; c < nchunks; ++c) {
      const int base = c * 512 + lane * 8;
      if (base < sk) {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float e = (v[c * 8 + j] == -INFINITY) ? 0.f
                                                : __expf(v[c * 8 + j] - mx);
          v[c * 8 + j] = e;
          sum += e;
        }
      }
    }
    sum = wave_reduce_sum(sum);
    const float inv = 1.f / sum;
#pragma unroll 4
    for (int c = 0; c < nchunks; ++c) {
      const int base = c * 512 + lane * 8;
      if (base < sk) {
        float o[8];
#pragma unroll
        for (int j = 0; j < 8; ++j) o[j] = v[c * 8 + j] * inv;
        store8<T>(yr + base, o);
      }
    }
  }
}

template <typename T>
__global__ void scaled_softmax_bwd_wave_kernel(
    const T* __restrict__ gy, const T* __restrict__ y, T* __restrict__ gx,
    float scale, long rows, int sk) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int nchunks = (sk + 511) / 512;
  float yv[32], gv[32];
  for (long row = (long)blockIdx.x * 4 + wid; row < rows;
       row += (long)gridDim.x * 4) {
    const T* yr = y + row * sk;
    const T* gr = gy + row * sk;
    T* gxr = gx + row * sk;
    float dot = 0.f;
#pragma unroll 4
    for (int c = 0; c < nchunks; ++c) {
      const int base = c * 512 + lane * 8;
      if (base < sk) {
        float ty[8], tg[8];
        load8<T>(yr + base, ty);
        load8<T>(gr + base, tg);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          yv[c * 8 + j] = ty[j];
          gv[c * 8 + j] = tg[j];
          dot += ty[j] * tg[j];
        }
      }
    }
    dot = wave_reduce_sum(dot);
#pragma unroll 4
    for (int c = 0; c < nchunks; ++c) {
      const int base = c * 512 + lane * 8;
      if (base < sk) {
        float o[8];
#pragma unroll
        for (int j = 0; j < 8; ++j)
          o[j] = scale * yv[c * 8 + j] * (gv[c * 8 + j] - dot);
        store8<T>(gxr + base, o);
      }
    }
  }
}

// ===========================================================================
// extern "C" launchers
// ===========================================================================
static inline int grid_for(long work_items, int block = 256) {
  long g = (work_items + block - 1) / block;
  if (g > FS_MAX_BLOCKS) g = FS_MAX_BLOCKS;
  if (g < 1) g = 1;
  return (int)g;
}

#define DISPATCH_DTYPE(dtype, T, ...)            \
  switch (dtype) {                               \
    case FS_F32: { using T = float; __VA_ARGS__; break; }   \
    case FS_BF16: { using T = bf16_t; __VA_ARGS__; break; } \
    case FS_F16: { using T = fp16_t; __VA_ARGS__; break; }  \
  }

extern "C" {

void fs_rms_norm_fwd(const void* x, const void* w, void* out, float* invrms,
                     int rows, int H, float eps, int dtype, hipStream_t s) {
  int grid = rows < FS_MAX_BLOCKS ? rows : FS_MAX_BLOCKS;
  size_t lds = (size_t)H * 4;
  DISPATCH_DTYPE(dtype, T,
    hipLaunchKernelGGL((rms_norm_fwd_kernel<T>), dim3(grid), dim3(256), lds, s,
                       (const T*)x, (const T*)w, (T*)out, invrms, rows, H, eps));
}

void fs_rms_norm_bwd(const void* gy, const void* x, const void* w,
                     const float* invrms, void* gx, float* gw_f32, int rows,
                     int H, int dtype, hipStream_t s) {
  int grid = rows < FS_MAX_BLOCKS ? rows : FS_MAX_BLOCKS;
  size_t lds = (size_t)H * 12;
  DISPATCH_DTYPE(dtype, T,
    hipLaunchKernelGGL((rms_norm_bwd_kernel<T>), dim3(grid), dim3(256), lds, s,
                       (const T*)gy, (const T*)x, (const T*)w, invrms, (T*)gx,
                       gw_f32, rows, H));
}

void fs_layer_norm_fwd(const void* x, const void* w, const void* b, void* out,
                       float* mean, float* invstd, int rows, int H, float eps,
                       int dtype, hipStream_t s) {
  int grid = rows < FS_MAX_BLOCKS ? rows : FS_MAX_BLOCKS;
  size_t lds = (size_t)H * 4;
  DISPATCH_DTYPE(dtype, T,
    hipLaunchKernelGGL((layer_norm_fwd_kernel<T>), dim3(grid), dim3(256), lds, s,
                       (const T*)x, (const T*)w, (const T*)b, (T*)out, mean,
                       invstd, rows, H, eps));
}

void fs_layer_norm_bwd(const void* gy, const void* x, const void* w,
                       const float* mean, const float* invstd, void* gx,
                       float* gw_f32, float* gb_f32, int rows, int H, int dtype,
                       hipStream_t s) {
  int grid = rows < FS_MAX_BLOCKS ? rows : FS_MAX_BLOCKS;
  size_t lds = (size_t)H * 16;
  DISPATCH_DTYPE(dtype, T,
    hipLaunchKernelGGL((layer_norm_bwd_kernel<T>), dim3(grid), dim3(256), lds, s,
                       (const T*)gy, (const T*)x, (const T*)w, mean, invstd,
                       (T*)gx, gw_f32, gb_f32, rows, H));
}

void fs_scaled_softmax_fwd(const void* x, const unsigned char* mask, void* out,
                           float scale, int rows, int sk, int sq, int np,
                           int mask_batches, int mask_sq, int mode, int dtype,
                           hipStream_t s) {
  if (sk <= 2048 && sk % 8 == 0) {
    // wave-per-row fast path (rows in registers)
    long wgrid = ((long)rows + 3) / 4;
    if (wgrid > 4096) wgrid = 4096;
    DISPATCH_DTYPE(dtype, T, {
      if (mode == 0)
        hipLaunchKernelGGL((scaled_softmax_fwd_wave_kernel<T, 0>),
                           dim3((unsigned)wgrid), dim3(256), 0, s,
                           (const T*)x, mask, (T*)out, scale, (long)rows, sk,
                           sq, np, mask_batches, mask_sq);
      else if (mode == 1)
        hipLaunchKernelGGL((scaled_softmax_fwd_wave_kernel<T, 1>),
                           dim3((unsigned)wgrid), dim3(256), 0, s,
                           (const T*)x, mask, (T*)out, scale, (long)rows, sk,
                           sq, np, mask_batches, mask_sq);
      else
        hipLaunchKernelGGL((scaled_softmax_fwd_wave_kernel<T, 2>),
                           dim3((unsigned)wgrid), dim3(256), 0, s,
                           (const T*)x, mask, (T*)out, scale, (long)rows, sk,
                           sq, np, mask_batches, mask_sq);
    });
    return;
  }
  int grid = rows < FS_MAX_BLOCKS ? rows : FS_MAX_BLOCKS;
  size_t lds = (size_t)sk * 4;
  DISPATCH_DTYPE(dtype, T, {
    if (mode == 0)
      hipLaunchKernelGGL((scaled_softmax_fwd_kernel<T, 0>), dim3(grid),
                         dim3(256), lds, s, (const T*)x, mask, (T*)out, scale,
                         rows, sk, sq, np, mask_batches, mask_sq);
    else if (mode == 1)
      hipLaunchKernelGGL((scaled_softmax_fwd_kernel<T, 1>), dim3(grid),
                         dim3(256), lds, s, (const T*)x, mask, (T*)out, scale,
                         rows, sk, sq, np, mask_batches, mask_sq);
    else
      hipLaunchKernelGGL((scaled_softmax_fwd_kernel<T, 2>), dim3(grid),
                         dim3(256), lds, s, (const T*)x, mask, (T*)out, scale,
                         rows, sk, sq, np, mask_batches, mask_sq);
  });
}

void fs_scaled_softmax_bwd(const void* gy, const void* y, void* gx, float scale,
                           int rows, int sk, int dtype, hipStream_t s) {
  if (sk <= 2048 && sk % 8 == 0) {
    long wgrid = ((long)rows + 3) / 4;
    if (wgrid > 4096) wgrid = 4096;
    DISPATCH_DTYPE(dtype, T,
      hipLaunchKernelGGL((scaled_softmax_bwd_wave_kernel<T>),
                         dim3((unsigned)wgrid), dim3(256), 0, s,
                         (const T*)gy, (const T*)y, (T*)gx, scale, (long)rows,
                         sk));
    return;
  }
  int grid = rows < FS_MAX_BLOCKS ? rows : FS_MAX_BLOCKS;
  size_t lds = (size_t)sk * 8;
  DISPATCH_DTYPE(dtype, T,
    hipLaunchKernelGGL((scaled_softmax_bwd_kernel<T>), dim3(grid), dim3(256),
                       lds, s, (const T*)gy, (const T*)y, (T*)gx, scale, rows,
                       sk));
}

void fs_rope(const void* x, void* out, const float* cos_t, const float* sin_t,
             long bn, int seq, int hn, int offset, int bwd, int dtype,
             hipStream_t s) {
  const int h2 = hn / 2;
  const long total_pairs = bn * (long)seq * h2;
  int grid = grid_for(total_pairs / 8);
  DISPATCH_DTYPE(dtype, T, {
    if (bwd)
      hipLaunchKernelGGL((rope_kernel<T, true>), dim3(grid), dim3(256), 0, s,
                         (const T*)x, (T*)out, cos_t, sin_t, total_pairs, seq,
                         h2, offset);
    else
      hipLaunchKernelGGL((rope_kernel<T, false>), dim3(grid), dim3(256), 0, s,
                         (const T*)x, (T*)out, cos_t, sin_t, total_pairs, seq,
                         h2, offset);
  });
}

void fs_swiglu_fwd(const void* packed, void* out, long rows, int h, int dtype,
                   hipStream_t s) {
  int grid = grid_for(rows * (long)h / 8);
  DISPATCH_DTYPE(dtype, T,
    hipLaunchKernelGGL((swiglu_fwd_kernel<T>), dim3(grid), dim3(256), 0, s,
                       (const T*)packed, (T*)out, rows, h));
}

void fs_swiglu_bwd(const void* gy, const void* packed, void* gpacked, long rows,
                   int h, int dtype, hipStream_t s) {
  int grid = grid_for(rows * (long)h / 8);
  DISPATCH_DTYPE(dtype, T,
    hipLaunchKernelGGL((swiglu_bwd_kernel<T>), dim3(grid), dim3(256), 0, s,
                       (const T*)gy, (const T*)packed, (T*)gpacked, rows, h));
}

void fs_bias_gelu(const void* x, const void* bias, const void* gy, void* out,
                  long rows, int h, int bwd, int dtype, hipStream_t s) {
  int grid = grid_for(rows * (long)h / 8);
  DISPATCH_DTYPE(dtype, T, {
    if (bwd)
      hipLaunchKernelGGL((bias_gelu_kernel<T, true>), dim3(grid), dim3(256), 0,
                         s, (const T*)x, (const T*)bias, (const T*)gy, (T*)out,
                         rows, h);
    else
      hipLaunchKernelGGL((bias_gelu_kernel<T, false>), dim3(grid), dim3(256), 0,
                         s, (const T*)x, (const T*)bias, (const T*)gy, (T*)out,
                         rows, h);
  });
}

void fs_fused_adamw(float* master, const void* grad, float* m, float* v,
                    void* out_param, long n, float lr, float beta1, float beta2,
                    float eps, float wd, int step, int grad_dtype,
                    int out_dtype, hipStream_t s) {
  const float bc1 = 1.f - powf(beta1, (float)step);
  const float bc2 = 1.f - powf(beta2, (float)step);
  int grid = grid_for((n + 3) / 4);
  DISPATCH_DTYPE(grad_dtype, GT, {
    DISPATCH_DTYPE(out_dtype, OT, {
      hipLaunchKernelGGL((fused_adamw_kernel<GT, OT>), dim3(grid), dim3(256), 0,
                         s, master, (const GT*)grad, m, v, (OT*)out_param, n,
                         lr, beta1, beta2, eps, wd, bc1, bc2);
    });
  });
}

}  // extern "C"

// ===========================================================================
// W8 GEMV: y[b, o] = scale[o] * sum_i q8[o, i] * x[b, i]   (decode path)
// Weight-bandwidth-bound: int8 weights halve HBM traffic vs bf16.
// One wave per output row; lanes stride the K dim with 16-byte int8 loads.
// ===========================================================================
typedef char char16_t_v __attribute__((ext_vector_type(16)));

__global__ __launch_bounds__(256)
void w8_gemv_kernel(const signed char* __restrict__ q8,
                    const float* __restrict__ scale,
                    const bf16_t* __restrict__ x,
                    bf16_t* __restrict__ y,
                    int batch, int in_features, int out_features) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int row0 = blockIdx.x * 4 + wid;
  for (int row = row0; row < out_features; row += gridDim.x * 4) {
    const signed char* wr = q8 + (long)row * in_features;
    for (int b = 0; b < batch; ++b) {
      const bf16_t* xb = x + (long)b * in_features;
      float acc = 0.f;
      for (int i = lane * 16; i + 15 < in_features; i += 64 * 16) {
        char16_t_v w = *reinterpret_cast<const char16_t_v*>(wr + i);
        float xv[8], xv2[8];
        load8<bf16_t>(xb + i, xv);
        load8<bf16_t>(xb + i + 8, xv2);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          acc += (float)w[j] * xv[j];
          acc += (float)w[j + 8] * xv2[j];
        }
      }
      acc = wave_reduce_sum(acc);
      if (lane == 0) y[(long)b * out_features + row] =
          from_f32<bf16_t>(acc * scale[row]);
    }
  }
}

extern "C" void fs_w8_gemv(const void* q8, const float* scale, const void* x,
                           void* y, int batch, int in_features,
                           int out_features, hipStream_t s) {
  int grid = (out_features + 3) / 4;
  if (grid > FS_MAX_BLOCKS) grid = FS_MAX_BLOCKS;
  hipLaunchKernelGGL(w8_gemv_kernel, dim3(grid), dim3(256), 0, s,
                     (const signed char*)q8, scale, (const bf16_t*)x,
                     (bf16_t*)y, batch, in_features, out_features);
}


// ===========================================================================
// Fused residual-add + RMSNorm (decode/inference): sum = a + b written
// out for the next residual, y = rmsnorm(sum) * w.  One launch instead
// of add + norm (the captured 13B decode graph carries 80 such pairs
// per token).  Inference-only: no invrms saved.
// ===========================================================================
__global__ void add_rms_norm_kernel(const bf16_t* __restrict__ a,
                                    const bf16_t* __restrict__ b,
                                    const bf16_t* __restrict__ w,
                                    bf16_t* __restrict__ sum_out,
                                    bf16_t* __restrict__ y,
                                    int rows, int H, float eps) {
  extern __shared__ float lds[];  // [H]
  __shared__ float red[32];
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const bf16_t* ar = a + (long)row * H;
    const bf16_t* br = b + (long)row * H;
    float ss = 0.f;
    for (int c = threadIdx.x * 8; c < H; c += blockDim.x * 8) {
      float va[8], vb[8], o[8];
      load8<bf16_t>(ar + c, va);
      load8<bf16_t>(br + c, vb);
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        o[i] = va[i] + vb[i];
        lds[c + i] = o[i];
        ss += o[i] * o[i];
      }
      store8<bf16_t>(sum_out + (long)row * H + c, o);
    }
    ss = block_reduce_sum(ss, red);
    const float ir = rsqrtf(ss / H + eps);
    for (int c = threadIdx.x * 8; c < H; c += blockDim.x * 8) {
      float wv[8], o[8];
      load8<bf16_t>(w + c, wv);
#pragma unroll
      for (int i = 0; i < 8; ++i) o[i] = lds[c + i] * ir * wv[i];
      store8<bf16_t>(y + (long)row * H + c, o);
    }
    __syncthreads();
  }
}

extern "C" void fs_add_rms_norm(const void* a, const void* b, const void* w,
                                void* sum_out, void* y, int rows, int H,
                                float eps, hipStream_t s) {
  int grid = rows < FS_MAX_BLOCKS ? rows : FS_MAX_BLOCKS;
  size_t lds = (size_t)H * 4;
  hipLaunchKernelGGL(add_rms_norm_kernel, dim3(grid), dim3(256), lds, s,
                     (const bf16_t*)a, (const bf16_t*)b, (const bf16_t*)w,
                     (bf16_t*)sum_out, (bf16_t*)y, rows, H, eps);
}

// ===========================================================================
// bf16 GEMV: y[b, o] = sum_i w[o, i] * x[b, i]   (bf16 decode path)
// hipBLASLt's M=1 GEMM leaves most of HBM bandwidth on the table on
// gfx950 (tile quantization: tiny M forces skinny tiles).  Decode is a
// pure weight stream — one wave per output row, lanes stride K with
// 32-byte loads, and a compile-time batch B<=8 reuses each weight dword
// for every sequence in the batch.
// ===========================================================================

template <int B>
__global__ __launch_bounds__(256)
void bf16_gemv_kernel(const bf16_t* __restrict__ w,
                      const bf16_t* __restrict__ x,
                      bf16_t* __restrict__ y,
                      int in_features, int out_features) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  for (int row = blockIdx.x * 4 + wid; row < out_features;
       row += gridDim.x * 4) {
    const bf16_t* wr = w + (long)row * in_features;
    float acc[B];
#pragma unroll
    for (int b = 0; b < B; ++b) acc[b] = 0.f;
    for (int i = lane * 16; i + 15 < in_features; i += 64 * 16) {
      float wv[8], wv2[8];
      load8<bf16_t>(wr + i, wv);
      load8<bf16_t>(wr + i + 8, wv2);
#pragma unroll
      for (int b = 0; b < B; ++b) {
        const bf16_t* xb = x + (long)b * in_features;
        float xv[8], xv2[8];
        load8<bf16_t>(xb + i, xv);
        load8<bf16_t>(xb + i + 8, xv2);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          acc[b] += wv[j] * xv[j] + wv2[j] * xv2[j];
      }
    }
#pragma unroll
    for (int b = 0; b < B; ++b) {
      float r = wave_reduce_sum(acc[b]);
      if (lane == 0)
        y[(long)b * out_features + row] = from_f32<bf16_t>(r);
    }
  }
}

extern "C" void fs_bf16_gemv(const void* w, const void* x, void* y, int batch,
                             int in_features, int out_features,
                             hipStream_t s) {
  int grid = (out_features + 3) / 4;
  if (grid > FS_MAX_BLOCKS) grid = FS_MAX_BLOCKS;
#define FS_GEMV_CASE(B)                                                      \
  case B:                                                                    \
    hipLaunchKernelGGL((bf16_gemv_kernel<B>), dim3(grid), dim3(256), 0, s,   \
                       (const bf16_t*)w, (const bf16_t*)x, (bf16_t*)y,       \
                       in_features, out_features);                           \
    break;
  switch (batch) {
    FS_GEMV_CASE(1) FS_GEMV_CASE(2) FS_GEMV_CASE(3) FS_GEMV_CASE(4)
    FS_GEMV_CASE(5) FS_GEMV_CASE(6) FS_GEMV_CASE(7) FS_GEMV_CASE(8)
  }
#undef FS_GEMV_CASE
}

// ===========================================================================
// Fused single-query decode attention (serving path)
// ===========================================================================
// One launch per layer replaces ~10 eager ops inside the captured decode
// graph: rope(q,k) at pos -> KV-cache write -> online-softmax attention
// over the cache prefix [0..pos] -> context vector.  The eager tail
// (rotate-half cats, fp32 casts of the cache, index_copy, softmax) was
// ~40% of the 13B per-token time (profiles/decprof).
//   qkv    [b, 3*H]     H = nh*D, straight out of the fused qkv GEMV
//   kc/vc  [b, nh, L, D] static caches, row `pos` written here
//   cos/sin [max_len, D] fp32 rope tables; pos = device scalar (long)
//   ctx    [b, H]
// Grid (b*nh) x 128 threads.  K tiles staged in LDS as fp32 rows padded
// to D+1 (bank-conflict-free per-thread row reads); V read coalesced by
// the D lanes in the accumulate phase.

#define FS_DEC_BLK 128

template <int D, bool ROPE>
__global__ __launch_bounds__(FS_DEC_BLK)
void decode_attn_kernel(const bf16_t* __restrict__ qkv,
                        bf16_t* __restrict__ kc, bf16_t* __restrict__ vc,
                        const float* __restrict__ cos_t,
                        const float* __restrict__ sin_t,
                        const long* __restrict__ pos_p,
                        bf16_t* __restrict__ ctx,
                        int nh, int cache_len, float scale) {
  const int tid = threadIdx.x;
  const int bh = blockIdx.x;  // b*nh + h
  const int b = bh / nh, h = bh % nh;
  const int pos = (int)pos_p[0];
  const int H = nh * D;

  __shared__ float qraw[D], kraw[D];
  __shared__ float qs[D], ks_new[D];
  __shared__ float red[16];
  __shared__ float pbuf[FS_DEC_BLK];
  __shared__ float ktile[FS_DEC_BLK][D + 1];

  // ---- phase 0: rope q/k at pos, write the cache row -------------------
  float v_new = 0.f;
  if (tid < D) {
    qraw[tid] = to_f32<bf16_t>(qkv[(long)b * 3 * H + h * D + tid]);
    kraw[tid] = to_f32<bf16_t>(qkv[(long)b * 3 * H + H + h * D + tid]);
    v_new = to_f32<bf16_t>(qkv[(long)b * 3 * H + 2 * H + h * D + tid]);
  }
  __syncthreads();
  if (tid < D) {
    float q = qraw[tid], k = kraw[tid];
    if (ROPE) {
      float c = cos_t[(long)pos * D + tid];
      float s = sin_t[(long)pos * D + tid];
      float qr = (tid < D / 2) ? -qraw[tid + D / 2] : qraw[tid - D / 2];
      float kr = (tid < D / 2) ? -kraw[tid + D / 2] : kraw[tid - D / 2];
      q = q * c + qr * s;
      k = k * c + kr * s;
    }
    qs[tid] = q * scale;
    ks_new[tid] = k;
    const long coff = (((long)b * nh + h) * cache_len + pos) * D + tid;
    kc[coff] = from_f32<bf16_t>(k);
    vc[coff] = from_f32<bf16_t>(v_new);
  }
  // make the cache row visible to this block's phase-2 V reads
  __threadfence();
  __syncthreads();

  // ---- phases 1+2: online softmax over chunks of 128 positions ---------
  const bf16_t* kbase = kc + ((long)b * nh + h) * (long)cache_len * D;
  const bf16_t* vbase = vc + ((long)b * nh + h) * (long)cache_len * D;
  float m_run = -INFINITY, l_run = 0.f, oacc = 0.f;
  for (int p0 = 0; p0 <= pos; p0 += FS_DEC_BLK) {
    const int nrow = min(FS_DEC_BLK, pos + 1 - p0);
    // stage K rows [p0, p0+nrow) -> fp32 LDS (coalesced 16B global loads)
    for (int idx = tid * 8; idx < nrow * D; idx += FS_DEC_BLK * 8) {
      const int r = idx / D, c = idx % D;  // D % 8 == 0: no row straddle
      float kv[8];
      load8<bf16_t>(kbase + (long)(p0 + r) * D + c, kv);
#pragma unroll
      for (int j = 0; j < 8; ++j) ktile[r][c + j] = kv[j];
    }
    __syncthreads();
    // the row written this step comes from LDS (no coherence question)
    if (pos >= p0 && pos < p0 + FS_DEC_BLK && tid < D)
      ktile[pos - p0][tid] = ks_new[tid];
    __syncthreads();
    // scores: thread t owns position p0+t
    float s_t = -INFINITY;
    if (tid < nrow) {
      float acc = 0.f;
#pragma unroll
      for (int j = 0; j < D; j += 8) {
#pragma unroll
        for (int jj = 0; jj < 8; ++jj)
          acc += qs[j + jj] * ktile[tid][j + jj];
      }
      s_t = acc;  // scale folded into qs
    }
    const float m_chunk = block_reduce_max(s_t, red);
    const float m_new = fmaxf(m_run, m_chunk);
    const float alpha = __expf(m_run - m_new);
    const float pr = (tid < nrow) ? __expf(s_t - m_new) : 0.f;
    pbuf[tid] = pr;
    const float psum = block_reduce_sum(pr, red);  // also syncs pbuf
    l_run = l_run * alpha + psum;
    m_run = m_new;
    // context accumulate: thread tid < D owns ctx element tid
    if (tid < D) {
      float acc = oacc * alpha;
      for (int t = 0; t < nrow; ++t)
        acc += pbuf[t] * to_f32<bf16_t>(vbase[(long)(p0 + t) * D + tid]);
      oacc = acc;
    }
    __syncthreads();  // pbuf/ktile reuse in next chunk
  }
  if (tid < D)
    ctx[(long)b * H + h * D + tid] = from_f32<bf16_t>(oacc / l_run);
}

extern "C" void fs_decode_attn(const void* qkv, void* kc, void* vc,
                               const float* cos_t, const float* sin_t,
                               const void* pos, void* ctx, int b, int nh,
                               int d, int cache_len, float scale, int rope,
                               hipStream_t s) {
#define FS_DEC_CASE(D)                                                       \
  if (d == D) {                                                              \
    if (rope)                                                                \
      hipLaunchKernelGGL((decode_attn_kernel<D, true>), dim3(b * nh),        \
                         dim3(FS_DEC_BLK), 0, s, (const bf16_t*)qkv,         \
                         (bf16_t*)kc, (bf16_t*)vc, cos_t, sin_t,             \
                         (const long*)pos, (bf16_t*)ctx, nh, cache_len,      \
                         scale);                                             \
    else                                                                     \
      hipLaunchKernelGGL((decode_attn_kernel<D, false>), dim3(b * nh),       \
                         dim3(FS_DEC_BLK), 0, s, (const bf16_t*)qkv,         \
                         (bf16_t*)kc, (bf16_t*)vc, cos_t, sin_t,             \
                         (const long*)pos, (bf16_t*)ctx, nh, cache_len,      \
                         scale);                                             \
    return;                                                                  \
  }
  FS_DEC_CASE(64)
  FS_DEC_CASE(128)
#undef FS_DEC_CASE
}

// ===========================================================================
// Fused vocab cross-entropy (TP-shard-aware)
// ===========================================================================
// Replaces the composite vocab_parallel_cross_entropy hot path, which
// materializes (and saves for backward) a full fp32 softmax over
// [N, V/tp] — 5.2 GB at the 13B bench shape.  The fused pair saves only
// per-row (local max, local sumexp(rel), predicted-logit) fp32 vectors;
// backward recomputes the softmax from the bf16 logits in one pass and
// emits bf16 grads.  Cross-shard reduction (max/sum/pred) stays on the
// torch side so the TP>1 all-reduce structure is unchanged.

__global__ __launch_bounds__(256)
void vocab_ce_fwd_kernel(const bf16_t* __restrict__ logits,
                         const long* __restrict__ targets,
                         float* __restrict__ m_out,
                         float* __restrict__ z_out,
                         float* __restrict__ pred_out,
                         long n, int w, int vstart, int vend) {
  __shared__ float red[8];
  for (long row = blockIdx.x; row < n; row += gridDim.x) {
    const bf16_t* lp = logits + row * w;
    // pass 1: row max (row stays L2-resident for pass 2)
    float m = -INFINITY;
    for (int j = threadIdx.x * 8; j < w; j += 256 * 8) {
      short8_t v8 = *reinterpret_cast<const short8_t*>(lp + j);
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        unsigned short u = (unsigned short)v8[k];
        m = fmaxf(m, __uint_as_float(((unsigned int)u) << 16));
      }
    }
    m = block_reduce_max(m, red);
    __syncthreads();  // red[] reuse hazard between reductions
    // pass 2: sumexp relative to the local max
    float z = 0.f;
    for (int j = threadIdx.x * 8; j < w; j += 256 * 8) {
      short8_t v8 = *reinterpret_cast<const short8_t*>(lp + j);
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        unsigned short u = (unsigned short)v8[k];
        z += __expf(__uint_as_float(((unsigned int)u) << 16) - m);
      }
    }
    z = block_reduce_sum(z, red);
    if (threadIdx.x == 0) {
      m_out[row] = m;
      z_out[row] = z;
      const long t = targets[row];
      float pred = 0.f;
      if (t >= vstart && t < vend)
        pred = __bfloat162float(lp[t - vstart]);
      pred_out[row] = pred;
    }
    __syncthreads();
  }
}

// dlogit[j] = (exp(l - M) / Z - onehot) * g   (M/Z are the GLOBAL row
// stats after the torch-side TP reduction)
__global__ __launch_bounds__(256)
void vocab_ce_bwd_kernel(const bf16_t* __restrict__ logits,
                         const long* __restrict__ targets,
                         const float* __restrict__ m_in,
                         const float* __restrict__ z_in,
                         const float* __restrict__ gout,
                         bf16_t* __restrict__ dlogits,
                         long n, int w, int vstart, int vend) {
  for (long row = blockIdx.x; row < n; row += gridDim.x) {
    const bf16_t* lp = logits + row * w;
    bf16_t* gp = dlogits + row * w;
    const float m = m_in[row];
    const float inv_z = 1.f / z_in[row];
    const float g = gout[row];
    const long t = targets[row] - vstart;
    for (int j = threadIdx.x * 8; j < w; j += 256 * 8) {
      short8_t v8 = *reinterpret_cast<const short8_t*>(lp + j);
      short8_t o8;
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        unsigned short u = (unsigned short)v8[k];
        float d = __expf(__uint_as_float(((unsigned int)u) << 16) - m)
                  * inv_z;
        if (j + k == t) d -= 1.f;
        o8[k] = (short)__hip_bfloat16_raw(__float2bfloat16(d * g)).x;
      }
      *reinterpret_cast<short8_t*>(gp + j) = o8;
    }
  }
}

extern "C" void fs_vocab_ce_fwd(const void* logits, const long* targets,
                                float* m_out, float* z_out, float* pred_out,
                                long n, int w, int vstart, int vend,
                                hipStream_t stream) {
  long blocks = n < FS_MAX_BLOCKS ? n : FS_MAX_BLOCKS;
  hipLaunchKernelGGL(vocab_ce_fwd_kernel, dim3((unsigned)blocks), dim3(256),
                     0, stream, (const bf16_t*)logits, targets, m_out, z_out,
                     pred_out, n, w, vstart, vend);
}

extern "C" void fs_vocab_ce_bwd(const void* logits, const long* targets,
                                const float* m_in, const float* z_in,
                                const float* gout, void* dlogits, long n,
                                int w, int vstart, int vend,
                                hipStream_t stream) {
  long blocks = n < FS_MAX_BLOCKS ? n : FS_MAX_BLOCKS;
  hipLaunchKernelGGL(vocab_ce_bwd_kernel, dim3((unsigned)blocks), dim3(256),
                     0, stream, (const bf16_t*)logits, targets, m_in, z_in,
                     gout, (bf16_t*)dlogits, n, w, vstart, vend);
}
