// Fused elementwise kernels: bias+GELU (K8 epilogue), SwiGLU gate (K9),
// RoPE rotate-half application (K5).

#include "common.h"

#define EW_BLOCK 256
// shadow accumulators for the column-reduction backward kernels: per-address
// atomic RMW chains shrink by this factor (outputs are [EW_SHADOWS, D],
// summed by the binding)
#define EW_SHADOWS 32

// fast tanh via the hardware exp pipe: tanh(y) = 1 - 2/(1 + exp(2y)).
// tanhf() is a slow polyline in libm; __expf is one v_exp_f32 — the GELU
// kernels were VALU-bound on tanhf in the ViT-L profile.
DEV_INLINE float fast_tanh(float y) {
  return 1.0f - 2.0f / (1.0f + __expf(2.0f * y));
}

DEV_INLINE float gelu_tanh(float x) {
  // tanh approximation (matches torch F.gelu(approximate="tanh"))
  const float k0 = 0.7978845608028654f;  // sqrt(2/pi)
  const float k1 = 0.044715f;
  float inner = k0 * (x + k1 * x * x * x);
  return 0.5f * x * (1.0f + fast_tanh(inner));
}

DEV_INLINE float gelu_tanh_grad(float x) {
  const float k0 = 0.7978845608028654f;
  const float k1 = 0.044715f;
  float x2 = x * x;
  float inner = k0 * (x + k1 * x * x2);
  float t = fast_tanh(inner);
  float sech2 = 1.0f - t * t;
  return 0.5f * (1.0f + t) + 0.5f * x * sech2 * k0 * (1.0f + 3.0f * k1 * x2);
}

// ---------------------------- bias + gelu ------------------------------
// Vectorized 8-wide (H % 8 == 0 — every FFN hidden dim). Backward assigns
// each thread a FIXED 8-column slice and walks rows, accumulating dbias in
// registers; one atomicAdd per column per thread at the end (guideline 12).

template <typename T>
__global__ void bias_gelu_fwd_kernel(const T* __restrict__ x, const T* __restrict__ bias,
                                     T* __restrict__ y, long rows, int H) {
  const long total8 = rows * (long)(H / 8);
  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < total8;
       idx += (long)gridDim.x * blockDim.x) {
    const int col8 = (int)(idx % (H / 8)) * 8;
    const T* xp = x + idx * 8;
    T* yp = y + idx * 8;
    T xb[8], bb[8], yb[8];
    Vec8<T>::load(xb, xp);
    Vec8<T>::load(bb, bias + col8);
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      float v = ScalarOps<T>::load(xb + e) + ScalarOps<T>::load(bb + e);
      ScalarOps<T>::store(yb + e, gelu_tanh(v));
    }
    Vec8<T>::store(yp, yb);
  }
}

template <typename T, int RR_ILP = 4>
__global__ void bias_gelu_bwd_kernel(const T* __restrict__ dy, const T* __restrict__ x,
                                     const T* __restrict__ bias, T* __restrict__ dx,
                                     float* __restrict__ dbias, long rows, int H) {
  // blockIdx.y tiles columns (256 threads x 8 cols); blockIdx.x strides rows.
  const int col8 = (blockIdx.y * blockDim.x + threadIdx.x) * 8;
  if (col8 >= H) return;
  T bb[8];
  Vec8<T>::load(bb, bias + col8);
  float db[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
  // RR_ILP rows in flight per iteration: the fixed-column walk is a strided
  // stream, so batch the loads for ILP/prefetch depth
  long row = blockIdx.x * (long)RR_ILP;
  const long rstep = (long)gridDim.x * RR_ILP;
  for (; row + RR_ILP - 1 < rows; row += rstep) {
    T xb[RR_ILP][8], gb[RR_ILP][8], ob[RR_ILP][8];
#pragma unroll
    for (int rr = 0; rr < RR_ILP; ++rr) {
      const long off = (row + rr) * (long)H + col8;
      Vec8<T>::load(xb[rr], x + off);
      Vec8<T>::load(gb[rr], dy + off);
    }
#pragma unroll
    for (int rr = 0; rr < RR_ILP; ++rr) {
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        float v = ScalarOps<T>::load(xb[rr] + e) + ScalarOps<T>::load(bb + e);
        float g = ScalarOps<T>::load(gb[rr] + e) * gelu_tanh_grad(v);
        ScalarOps<T>::store(ob[rr] + e, g);
        db[e] += g;
      }
      Vec8<T>::store(dx + (row + rr) * (long)H + col8, ob[rr]);
    }
  }
  for (; row < rows; ++row) {
    const long off = row * (long)H + col8;
    T xb[8], gb[8], ob[8];
    Vec8<T>::load(xb, x + off);
    Vec8<T>::load(gb, dy + off);
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      float v = ScalarOps<T>::load(xb + e) + ScalarOps<T>::load(bb + e);
      float g = ScalarOps<T>::load(gb + e) * gelu_tanh_grad(v);
      ScalarOps<T>::store(ob + e, g);
      db[e] += g;
    }
    Vec8<T>::store(dx + off, ob);
  }
  float* dbias_s = dbias + (long)(blockIdx.x & (EW_SHADOWS - 1)) * H;
#pragma unroll
  for (int e = 0; e < 8; ++e) atomicAdd(dbias_s + col8 + e, db[e]);
}

// Contiguous-row bias+GELU backward (A/B variant, DINOV3_BG_ROWS=1): each
// block walks whole rows (PASSES x 2048-col sweeps, perfectly coalesced
// 4 KB bursts) instead of a fixed column slice's strided walk; dbias
// accumulates in registers per (thread, pass) column set and leaves via the
// shadowed atomics. PASSES = H / (256*8).
template <typename T, int PASSES>
__global__ __launch_bounds__(256) void bias_gelu_bwd_rows_kernel(
    const T* __restrict__ dy, const T* __restrict__ x, const T* __restrict__ bias,
    T* __restrict__ dx, float* __restrict__ dbias, long rows, int H) {
  T bb[PASSES][8];
  float db[PASSES][8];
#pragma unroll
  for (int pss = 0; pss < PASSES; ++pss) {
    Vec8<T>::load(bb[pss], bias + (pss * 256 + threadIdx.x) * 8);
#pragma unroll
    for (int e = 0; e < 8; ++e) db[pss][e] = 0.f;
  }
  for (long row = blockIdx.x; row < rows; row += gridDim.x) {
    const long base = row * (long)H;
#pragma unroll
    for (int pss = 0; pss < PASSES; ++pss) {
      const long off = base + (pss * 256 + threadIdx.x) * 8;
      T xb[8], gb[8], ob[8];
      Vec8<T>::load(xb, x + off);
      Vec8<T>::load(gb, dy + off);
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const float v = ScalarOps<T>::load(xb + e) + ScalarOps<T>::load(bb[pss] + e);
        const float g = ScalarOps<T>::load(gb + e) * gelu_tanh_grad(v);
        ScalarOps<T>::store(ob + e, g);
        db[pss][e] += g;
      }
      Vec8<T>::store(dx + off, ob);
    }
  }
  float* dbias_s = dbias + (long)(blockIdx.x & (EW_SHADOWS - 1)) * H;
#pragma unroll
  for (int pss = 0; pss < PASSES; ++pss) {
    const int c8 = (pss * 256 + threadIdx.x) * 8;
#pragma unroll
    for (int e = 0; e < 8; ++e) atomicAdd(dbias_s + c8 + e, db[pss][e]);
  }
}

// -------------------- LayerScale + residual add (K10) -------------------
// out = x + gamma * res ; bwd: dx = dout (aliased), dres = dout * gamma,
// dgamma = colsum(dout * res). 8-wide; same fixed-column-slice register
// accumulation as bias_gelu_bwd.

template <typename T>
__global__ void ls_axpy_fwd_kernel(const T* __restrict__ x, const T* __restrict__ res,
                                   const T* __restrict__ gamma, T* __restrict__ out,
                                   long rows, int D) {
  const long total8 = rows * (long)(D / 8);
  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < total8;
       idx += (long)gridDim.x * blockDim.x) {
    const int col8 = (int)(idx % (D / 8)) * 8;
    T xb[8], rb[8], gb[8], ob[8];
    Vec8<T>::load(xb, x + idx * 8);
    Vec8<T>::load(rb, res + idx * 8);
    Vec8<T>::load(gb, gamma + col8);
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      float v = ScalarOps<T>::load(xb + e) +
                ScalarOps<T>::load(gb + e) * ScalarOps<T>::load(rb + e);
      ScalarOps<T>::store(ob + e, v);
    }
    Vec8<T>::store(out + idx * 8, ob);
  }
}

template <typename T>
__global__ void ls_axpy_bwd_kernel(const T* __restrict__ dout, const T* __restrict__ res,
                                   const T* __restrict__ gamma, T* __restrict__ dres,
                                   float* __restrict__ dgamma, long rows, int D) {
  const int col8 = (blockIdx.y * blockDim.x + threadIdx.x) * 8;
  if (col8 >= D) return;
  T gb[8];
  Vec8<T>::load(gb, gamma + col8);
  float dg[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
  // 4 rows in flight: with the atomic-chain-capped grid (~384 blocks) a
  // one-row walk leaves too few loads outstanding to cover HBM latency
  long row = blockIdx.x * 4L;
  const long rstep = (long)gridDim.x * 4;
  for (; row + 3 < rows; row += rstep) {
    T db[4][8], rb[4][8], ob[4][8];
#pragma unroll
    for (int rr = 0; rr < 4; ++rr) {
      Vec8<T>::load(db[rr], dout + (row + rr) * (long)D + col8);
      Vec8<T>::load(rb[rr], res + (row + rr) * (long)D + col8);
    }
#pragma unroll
    for (int rr = 0; rr < 4; ++rr) {
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const float g = ScalarOps<T>::load(db[rr] + e);
        dg[e] += g * ScalarOps<T>::load(rb[rr] + e);
        ScalarOps<T>::store(ob[rr] + e, g * ScalarOps<T>::load(gb + e));
      }
      Vec8<T>::store(dres + (row + rr) * (long)D + col8, ob[rr]);
    }
  }
  for (; row < rows; ++row) {
    const long off = row * (long)D + col8;
    T db[8], rb[8], ob[8];
    Vec8<T>::load(db, dout + off);
    Vec8<T>::load(rb, res + off);
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      const float g = ScalarOps<T>::load(db + e);
      dg[e] += g * ScalarOps<T>::load(rb + e);
      ScalarOps<T>::store(ob + e, g * ScalarOps<T>::load(gb + e));
    }
    Vec8<T>::store(dres + off, ob);
  }
  float* dgamma_s = dgamma + (long)(blockIdx.x & (EW_SHADOWS - 1)) * D;
#pragma unroll
  for (int e = 0; e < 8; ++e) atomicAdd(dgamma_s + col8 + e, dg[e]);
}

// ---- bias-fused variants (DINOV3_FUSED_RESIDUAL): out = x + gamma*(res+bias).
// Separate kernels so the validated ls_axpy binaries above stay untouched.

template <typename T>
__global__ void ls_axpy_bias_fwd_kernel(const T* __restrict__ x, const T* __restrict__ res,
                                        const T* __restrict__ gamma,
                                        const T* __restrict__ bias, T* __restrict__ out,
                                        long rows, int D) {
  const long total8 = rows * (long)(D / 8);
  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < total8;
       idx += (long)gridDim.x * blockDim.x) {
    const int col8 = (int)(idx % (D / 8)) * 8;
    T xb[8], rb[8], gb[8], bb[8], ob[8];
    Vec8<T>::load(xb, x + idx * 8);
    Vec8<T>::load(rb, res + idx * 8);
    Vec8<T>::load(gb, gamma + col8);
    Vec8<T>::load(bb, bias + col8);
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      float v = ScalarOps<T>::load(xb + e) +
                ScalarOps<T>::load(gb + e) *
                    (ScalarOps<T>::load(rb + e) + ScalarOps<T>::load(bb + e));
      ScalarOps<T>::store(ob + e, v);
    }
    Vec8<T>::store(out + idx * 8, ob);
  }
}

template <typename T>
__global__ void ls_axpy_bias_bwd_kernel(const T* __restrict__ dout,
                                        const T* __restrict__ res,
                                        const T* __restrict__ gamma,
                                        const T* __restrict__ bias, T* __restrict__ dres,
                                        float* __restrict__ dgamma,
                                        float* __restrict__ dbias, long rows, int D) {
  const int col8 = (blockIdx.y * blockDim.x + threadIdx.x) * 8;
  if (col8 >= D) return;
  T gb[8], bb[8];
  Vec8<T>::load(gb, gamma + col8);
  Vec8<T>::load(bb, bias + col8);
  float dg[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
  float db_acc[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
  long row = blockIdx.x * 4L;
  const long rstep = (long)gridDim.x * 4;
  for (; row + 3 < rows; row += rstep) {
    T db[4][8], rb[4][8], ob[4][8];
#pragma unroll
    for (int rr = 0; rr < 4; ++rr) {
      Vec8<T>::load(db[rr], dout + (row + rr) * (long)D + col8);
      Vec8<T>::load(rb[rr], res + (row + rr) * (long)D + col8);
    }
#pragma unroll
    for (int rr = 0; rr < 4; ++rr) {
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const float g = ScalarOps<T>::load(db[rr] + e);
        const float gm = ScalarOps<T>::load(gb + e);
        dg[e] += g * (ScalarOps<T>::load(rb[rr] + e) + ScalarOps<T>::load(bb + e));
        db_acc[e] += g * gm;
        ScalarOps<T>::store(ob[rr] + e, g * gm);
      }
      Vec8<T>::store(dres + (row + rr) * (long)D + col8, ob[rr]);
    }
  }
  for (; row < rows; ++row) {
    const long off = row * (long)D + col8;
    T db[8], rb[8], ob[8];
    Vec8<T>::load(db, dout + off);
    Vec8<T>::load(rb, res + off);
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      const float g = ScalarOps<T>::load(db + e);
      const float gm = ScalarOps<T>::load(gb + e);
      dg[e] += g * (ScalarOps<T>::load(rb + e) + ScalarOps<T>::load(bb + e));
      db_acc[e] += g * gm;
      ScalarOps<T>::store(ob + e, g * gm);
    }
    Vec8<T>::store(dres + off, ob);
  }
  const long sh = (long)(blockIdx.x & (EW_SHADOWS - 1)) * D;
#pragma unroll
  for (int e = 0; e < 8; ++e) {
    atomicAdd(dgamma + sh + col8 + e, dg[e]);
    atomicAdd(dbias + sh + col8 + e, db_acc[e]);
  }
}

// ---------------- fused LayerScale scatter-add (K10 + K11) ----------------
// Stochastic-depth residual with the LayerScale gamma and the producing
// GEMM's bias folded in (kills the standalone gamma multiply and the bias
// epilogue + bias-grad reduction of the proj/fc2 Linears):
//   dst[idx[r], c] += scale[r] * gamma[c] * (src[r, c] + bias[c])
// Rows are disjoint (drop-path subsets), so no atomics on dst.

template <typename T>
__global__ void ls_scatter_add_kernel(T* __restrict__ dst, const long* __restrict__ idx,
                                      const T* __restrict__ src,
                                      const T* __restrict__ gamma,
                                      const T* __restrict__ bias,
                                      const float* __restrict__ scale, long M, int D) {
  const int d8 = D / 8;
  long total = M * (long)d8;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    const long r = i / d8;
    const int c8 = (int)(i % d8) * 8;
    const long drow = idx[r];
    const float s = scale != nullptr ? scale[r] : 1.0f;
    T sb[8], db[8], gb[8], bb[8];
    Vec8<T>::load(sb, src + r * (long)D + c8);
    Vec8<T>::load(db, dst + drow * (long)D + c8);
    if (gamma != nullptr) Vec8<T>::load(gb, gamma + c8);
    if (bias != nullptr) Vec8<T>::load(bb, bias + c8);
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      const float g = gamma != nullptr ? ScalarOps<T>::load(gb + e) : 1.0f;
      const float bv = bias != nullptr ? ScalarOps<T>::load(bb + e) : 0.0f;
      ScalarOps<T>::store(
          db + e, ScalarOps<T>::load(db + e) + s * g * (ScalarOps<T>::load(sb + e) + bv));
    }
    Vec8<T>::store(dst + drow * (long)D + c8, db);
  }
}

// backward: dres[r,c] = dy[idx[r],c] * gamma[c] * scale[r];
//           dgamma[c] += sum_r dy[idx[r],c] * (src[r,c]+bias[c]) * scale[r];
//           dbias[c]  += sum_r dy[idx[r],c] * gamma[c] * scale[r]
template <typename T, int RR = 4>
__global__ void ls_scatter_bwd_kernel(const T* __restrict__ dy, const long* __restrict__ idx,
                                      const T* __restrict__ src,
                                      const T* __restrict__ gamma,
                                      const T* __restrict__ bias,
                                      const float* __restrict__ scale,
                                      T* __restrict__ dres, float* __restrict__ dgamma,
                                      float* __restrict__ dbias, long M, int D) {
  const int col8 = (blockIdx.y * blockDim.x + threadIdx.x) * 8;
  if (col8 >= D) return;
  T gb[8], bb[8];
  if (gamma != nullptr) Vec8<T>::load(gb, gamma + col8);
  if (bias != nullptr) Vec8<T>::load(bb, bias + col8);
  float dg[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
  float db_acc[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
  long r = blockIdx.x * (long)RR;
  const long rstep4 = (long)gridDim.x * RR;
  for (; r + RR - 1 < M; r += rstep4) {
    long srow[RR];
    float sc[RR];
    T dyb[RR][8], rb[RR][8], ob[RR][8];
#pragma unroll
    for (int rr = 0; rr < RR; ++rr) {
      srow[rr] = idx[r + rr];
      sc[rr] = scale != nullptr ? scale[r + rr] : 1.0f;
      Vec8<T>::load(dyb[rr], dy + srow[rr] * (long)D + col8);
      Vec8<T>::load(rb[rr], src + (r + rr) * (long)D + col8);
    }
#pragma unroll
    for (int rr = 0; rr < RR; ++rr) {
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const float gv = ScalarOps<T>::load(dyb[rr] + e) * sc[rr];
        const float gm = gamma != nullptr ? ScalarOps<T>::load(gb + e) : 1.0f;
        const float bv = bias != nullptr ? ScalarOps<T>::load(bb + e) : 0.0f;
        dg[e] += gv * (ScalarOps<T>::load(rb[rr] + e) + bv);
        db_acc[e] += gv * gm;
        ScalarOps<T>::store(ob[rr] + e, gv * gm);
      }
      Vec8<T>::store(dres + (r + rr) * (long)D + col8, ob[rr]);
    }
  }
  for (; r < M; ++r) {
    const long srow = idx[r];
    const float s = scale != nullptr ? scale[r] : 1.0f;
    T dyb[8], rb[8], ob[8];
    Vec8<T>::load(dyb, dy + srow * (long)D + col8);
    Vec8<T>::load(rb, src + r * (long)D + col8);
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      const float gv = ScalarOps<T>::load(dyb + e) * s;
      const float gm = gamma != nullptr ? ScalarOps<T>::load(gb + e) : 1.0f;
      const float bv = bias != nullptr ? ScalarOps<T>::load(bb + e) : 0.0f;
      dg[e] += gv * (ScalarOps<T>::load(rb + e) + bv);
      db_acc[e] += gv * gm;
      ScalarOps<T>::store(ob + e, gv * gm);
    }
    Vec8<T>::store(dres + r * (long)D + col8, ob);
  }
  const long sh = (long)(blockIdx.x & (EW_SHADOWS - 1)) * D;
#pragma unroll
  for (int e = 0; e < 8; ++e) {
    if (dgamma != nullptr) atomicAdd(dgamma + sh + col8 + e, dg[e]);
    if (dbias != nullptr) atomicAdd(dbias + sh + col8 + e, db_acc[e]);
  }
}

// ------------------------------ swiglu ---------------------------------
// x12 = [rows, 2H] as [x1 | x2]; y = silu(x1) * x2.

template <typename T>
__global__ void swiglu_fwd_kernel(const T* __restrict__ x12, T* __restrict__ y,
                                  long rows, int H) {
  long total = rows * (long)H;
  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    long row = idx / H;
    int col = (int)(idx % H);
    const T* xr = x12 + row * (long)(2 * H);
    float x1 = ScalarOps<T>::load(xr + col);
    float x2 = ScalarOps<T>::load(xr + H + col);
    float sig = 1.0f / (1.0f + expf(-x1));
    ScalarOps<T>::store(y + idx, x1 * sig * x2);
  }
}

template <typename T>
__global__ void swiglu_bwd_kernel(const T* __restrict__ dy, const T* __restrict__ x12,
                                  T* __restrict__ dx12, long rows, int H) {
  long total = rows * (long)H;
  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    long row = idx / H;
    int col = (int)(idx % H);
    const T* xr = x12 + row * (long)(2 * H);
    T* dxr = dx12 + row * (long)(2 * H);
    float x1 = ScalarOps<T>::load(xr + col);
    float x2 = ScalarOps<T>::load(xr + H + col);
    float g = ScalarOps<T>::load(dy + idx);
    float sig = 1.0f / (1.0f + expf(-x1));
    float silu = x1 * sig;
    float dsilu = sig * (1.0f + x1 * (1.0f - sig));
    ScalarOps<T>::store(dxr + col, g * x2 * dsilu);
    ScalarOps<T>::store(dxr + H + col, g * silu);
  }
}

// -------------------- row gather / scatter-add (K11) --------------------
// Stochastic-depth subset compute: gather kept samples' token rows, compute,
// scatter-add the scaled residual back. Rows are disjoint (no atomics).

template <typename T>
__global__ void row_gather_kernel(const T* __restrict__ src, const long* __restrict__ idx,
                                  T* __restrict__ out, long M, int D) {
  const int d8 = D / 8;
  long total = M * (long)d8;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    const long r = i / d8;
    const int c8 = (int)(i % d8) * 8;
    const long srow = idx[r];
    T buf[8];
    Vec8<T>::load(buf, src + srow * (long)D + c8);
    Vec8<T>::store(out + r * (long)D + c8, buf);
  }
}

template <typename T>
__global__ void row_scatter_add_kernel(T* __restrict__ dst, const long* __restrict__ idx,
                                       const T* __restrict__ src,
                                       const float* __restrict__ scale, long M, int D) {
  const int d8 = D / 8;
  long total = M * (long)d8;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    const long r = i / d8;
    const int c8 = (int)(i % d8) * 8;
    const long drow = idx[r];
    const float s = scale != nullptr ? scale[r] : 1.0f;
    T sb[8], db[8];
    Vec8<T>::load(sb, src + r * (long)D + c8);
    Vec8<T>::load(db, dst + drow * (long)D + c8);
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      ScalarOps<T>::store(db + e,
                          ScalarOps<T>::load(db + e) + s * ScalarOps<T>::load(sb + e));
    }
    Vec8<T>::store(dst + drow * (long)D + c8, db);
  }
}

// gather with per-row scale (backward of scatter-add)
template <typename T>
__global__ void row_gather_scaled_kernel(const T* __restrict__ src,
                                         const long* __restrict__ idx,
                                         const float* __restrict__ scale,
                                         T* __restrict__ out, long M, int D) {
  const int d8 = D / 8;
  long total = M * (long)d8;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    const long r = i / d8;
    const int c8 = (int)(i % d8) * 8;
    const long srow = idx[r];
    const float s = scale != nullptr ? scale[r] : 1.0f;
    T buf[8];
    Vec8<T>::load(buf, src + srow * (long)D + c8);
#pragma unroll
    for (int e = 0; e < 8; ++e)
      ScalarOps<T>::store(buf + e, s * ScalarOps<T>::load(buf + e));
    Vec8<T>::store(out + r * (long)D + c8, buf);
  }
}

// ------------------------------- RoPE ----------------------------------
// x: [B, Hh, N, hd] contiguous; sin/cos: [P, hd] fp32 with P = N - prefix.
// rotate-half: out[j] = x[j]*cos[j] - x[j+hd/2]*sin[j]          (j < hd/2)
//              out[j] = x[j]*cos[j] + x[j-hd/2]*sin[j]          (j >= hd/2)
// Tokens [0, prefix) pass through unchanged (cls + storage tokens).
// One thread per pair (j < hd/2) handles both halves: 2 loads, 2 stores.

template <typename T>
__global__ void rope_fwd_kernel(const T* __restrict__ x, const float* __restrict__ sin_t,
                                const float* __restrict__ cos_t, T* __restrict__ y,
                                long BH, int N, int P, int hd) {
  const int half = hd / 2;
  long total = BH * (long)N * half;
  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    int j = (int)(idx % half);
    long t = idx / half;          // BH*N token index
    int n = (int)(t % N);
    long base = t * (long)hd;
    int p = n - (N - P);          // patch index (negative for prefix tokens)
    float lo = ScalarOps<T>::load(x + base + j);
    float hi = ScalarOps<T>::load(x + base + half + j);
    if (p < 0) {
      ScalarOps<T>::store(y + base + j, lo);
      ScalarOps<T>::store(y + base + half + j, hi);
    } else {
      const float* srow = sin_t + (long)p * hd;
      const float* crow = cos_t + (long)p * hd;
      float s_lo = srow[j], c_lo = crow[j];
      float s_hi = srow[half + j], c_hi = crow[half + j];
      ScalarOps<T>::store(y + base + j, lo * c_lo - hi * s_lo);
      ScalarOps<T>::store(y + base + half + j, hi * c_hi + lo * s_hi);
    }
  }
}

// ---------------------------- C wrappers -------------------------------

template <typename T>
void launch_bias_gelu_fwd(const T* x, const T* bias, T* y, long rows, int H,
                          hipStream_t stream) {
  long total8 = rows * (long)(H / 8);
  int grid = (int)min((total8 + EW_BLOCK - 1) / EW_BLOCK, (long)2048);
  hipLaunchKernelGGL((bias_gelu_fwd_kernel<T>), dim3(grid), dim3(EW_BLOCK), 0, stream,
                     x, bias, y, rows, H);
}

template <typename T>
void launch_bias_gelu_bwd(const T* dy, const T* x, const T* bias, T* dx, float* dbias,
                          long rows, int H, hipStream_t stream) {
  const int col_tiles = (H / 8 + EW_BLOCK - 1) / EW_BLOCK;
  static const int rowsv = [] {
    const char* e = getenv("DINOV3_BG_ROWS");
    return e && e[0] == '1';
  }();
  if (rowsv && H == 4096) {
    int grid = (int)min(rows, (long)2048);
    hipLaunchKernelGGL((bias_gelu_bwd_rows_kernel<T, 2>), dim3(grid), dim3(256), 0,
                       stream, dy, x, bias, dx, dbias, rows, H);
    return;
  }
  static const int ilp8 = [] {
    const char* e = getenv("DINOV3_BG_ILP8");
    return e && e[0] == '1';
  }();
  if (ilp8) {
    int row_grid = (int)min((rows + 7) / 8, (long)(2048 / col_tiles + 1));
    hipLaunchKernelGGL((bias_gelu_bwd_kernel<T, 8>), dim3(row_grid, col_tiles),
                       dim3(EW_BLOCK), 0, stream, dy, x, bias, dx, dbias, rows, H);
    return;
  }
  int row_grid = (int)min((rows + 3) / 4, (long)(2048 / col_tiles + 1));
  hipLaunchKernelGGL((bias_gelu_bwd_kernel<T, 4>), dim3(row_grid, col_tiles),
                     dim3(EW_BLOCK), 0, stream, dy, x, bias, dx, dbias, rows, H);
}

template <typename T>
void launch_ls_axpy_fwd(const T* x, const T* res, const T* gamma, T* out, long rows,
                        int D, hipStream_t stream) {
  long total8 = rows * (long)(D / 8);
  int grid = (int)min((total8 + EW_BLOCK - 1) / EW_BLOCK, (long)2048);
  hipLaunchKernelGGL((ls_axpy_fwd_kernel<T>), dim3(grid), dim3(EW_BLOCK), 0, stream, x,
                     res, gamma, out, rows, D);
}

template <typename T>
void launch_ls_axpy_bwd(const T* dout, const T* res, const T* gamma, T* dres,
                        float* dgamma, long rows, int D, hipStream_t stream) {
  // block sized to the row width (no idle half-waves at D=1024) and the
  // grid bounded: dgamma's per-column atomic chain depth == blocks/shadows
  const int block = D / 8 < EW_BLOCK ? D / 8 : EW_BLOCK;
  const int col_tiles = (D / 8 + block - 1) / block;
  int row_grid = (int)min((rows + 7) / 8, (long)(1024 / col_tiles + 1));
  hipLaunchKernelGGL((ls_axpy_bwd_kernel<T>), dim3(row_grid, col_tiles), dim3(block),
                     0, stream, dout, res, gamma, dres, dgamma, rows, D);
}

template <typename T>
void launch_ls_axpy_bias_fwd(const T* x, const T* res, const T* gamma, const T* bias,
                             T* out, long rows, int D, hipStream_t stream) {
  long total8 = rows * (long)(D / 8);
  int grid = (int)min((total8 + EW_BLOCK - 1) / EW_BLOCK, (long)2048);
  hipLaunchKernelGGL((ls_axpy_bias_fwd_kernel<T>), dim3(grid), dim3(EW_BLOCK), 0, stream,
                     x, res, gamma, bias, out, rows, D);
}

template <typename T>
void launch_ls_axpy_bias_bwd(const T* dout, const T* res, const T* gamma, const T* bias,
                             T* dres, float* dgamma, float* dbias, long rows, int D,
                             hipStream_t stream) {
  const int block = D / 8 < EW_BLOCK ? D / 8 : EW_BLOCK;
  const int col_tiles = (D / 8 + block - 1) / block;
  int row_grid = (int)min((rows + 7) / 8, (long)(1024 / col_tiles + 1));
  hipLaunchKernelGGL((ls_axpy_bias_bwd_kernel<T>), dim3(row_grid, col_tiles),
                     dim3(block), 0, stream, dout, res, gamma, bias, dres, dgamma,
                     dbias, rows, D);
}

template <typename T>
void launch_ls_scatter_add(T* dst, const long* idx, const T* src, const T* gamma,
                           const T* bias, const float* scale, long M, int D,
                           hipStream_t stream) {
  long total = M * (long)(D / 8);
  int grid = (int)min((total + EW_BLOCK - 1) / EW_BLOCK, (long)2048);
  if (grid == 0) return;
  hipLaunchKernelGGL((ls_scatter_add_kernel<T>), dim3(grid), dim3(EW_BLOCK), 0, stream,
                     dst, idx, src, gamma, bias, scale, M, D);
}

template <typename T>
void launch_ls_scatter_bwd(const T* dy, const long* idx, const T* src, const T* gamma,
                           const T* bias, const float* scale, T* dres, float* dgamma,
                           float* dbias, long M, int D, hipStream_t stream) {
  const int block = D / 8 < EW_BLOCK ? D / 8 : EW_BLOCK;
  const int col_tiles = (D / 8 + block - 1) / block;
  // 4 rows in flight (8 halves occupancy, r2_gpu10) over a 1024-block grid
  // (2048 measured 34% slower, r2_gpu13); shadows keep atomic chains short
  int row_grid = (int)min(M > 0 ? (M + 7) / 8 : 1, (long)(1024 / col_tiles + 1));
  hipLaunchKernelGGL((ls_scatter_bwd_kernel<T, 4>), dim3(row_grid, col_tiles), dim3(block),
                     0, stream, dy, idx, src, gamma, bias, scale, dres, dgamma, dbias, M,
                     D);
}

template <typename T>
void launch_row_gather(const T* src, const long* idx, T* out, long M, int D,
                       hipStream_t stream) {
  long total = M * (long)(D / 8);
  int grid = (int)min((total + EW_BLOCK - 1) / EW_BLOCK, (long)2048);
  if (grid == 0) return;
  hipLaunchKernelGGL((row_gather_kernel<T>), dim3(grid), dim3(EW_BLOCK), 0, stream, src,
                     idx, out, M, D);
}

template <typename T>
void launch_row_scatter_add(T* dst, const long* idx, const T* src, const float* scale,
                            long M, int D, hipStream_t stream) {
  long total = M * (long)(D / 8);
  int grid = (int)min((total + EW_BLOCK - 1) / EW_BLOCK, (long)2048);
  if (grid == 0) return;
  hipLaunchKernelGGL((row_scatter_add_kernel<T>), dim3(grid), dim3(EW_BLOCK), 0, stream,
                     dst, idx, src, scale, M, D);
}

template <typename T>
void launch_row_gather_scaled(const T* src, const long* idx, const float* scale, T* out,
                              long M, int D, hipStream_t stream) {
  long total = M * (long)(D / 8);
  int grid = (int)min((total + EW_BLOCK - 1) / EW_BLOCK, (long)2048);
  if (grid == 0) return;
  hipLaunchKernelGGL((row_gather_scaled_kernel<T>), dim3(grid), dim3(EW_BLOCK), 0, stream,
                     src, idx, scale, out, M, D);
}

template <typename T>
void launch_swiglu_fwd(const T* x12, T* y, long rows, int H, hipStream_t stream) {
  long total = rows * (long)H;
  int grid = (int)min((total + EW_BLOCK - 1) / EW_BLOCK, (long)2048);
  hipLaunchKernelGGL((swiglu_fwd_kernel<T>), dim3(grid), dim3(EW_BLOCK), 0, stream,
                     x12, y, rows, H);
}

template <typename T>
void launch_swiglu_bwd(const T* dy, const T* x12, T* dx12, long rows, int H,
                       hipStream_t stream) {
  long total = rows * (long)H;
  int grid = (int)min((total + EW_BLOCK - 1) / EW_BLOCK, (long)2048);
  hipLaunchKernelGGL((swiglu_bwd_kernel<T>), dim3(grid), dim3(EW_BLOCK), 0, stream,
                     dy, x12, dx12, rows, H);
}

template <typename T>
void launch_rope_fwd(const T* x, const float* sin_t, const float* cos_t, T* y, long BH,
                     int N, int P, int hd, hipStream_t stream) {
  long total = BH * (long)N * (hd / 2);
  int grid = (int)min((total + EW_BLOCK - 1) / EW_BLOCK, (long)4096);
  hipLaunchKernelGGL((rope_fwd_kernel<T>), dim3(grid), dim3(EW_BLOCK), 0, stream,
                     x, sin_t, cos_t, y, BH, N, P, hd);
}

#define INSTANTIATE_EW(T)                                                            \
  template void launch_bias_gelu_fwd<T>(const T*, const T*, T*, long, int,           \
                                        hipStream_t);                                \
  template void launch_ls_axpy_fwd<T>(const T*, const T*, const T*, T*, long, int,   \
                                      hipStream_t);                                  \
  template void launch_ls_axpy_bwd<T>(const T*, const T*, const T*, T*, float*,      \
                                      long, int, hipStream_t);                       \
  template void launch_ls_axpy_bias_fwd<T>(const T*, const T*, const T*, const T*,   \
                                           T*, long, int, hipStream_t);              \
  template void launch_ls_axpy_bias_bwd<T>(const T*, const T*, const T*, const T*,   \
                                           T*, float*, float*, long, int,            \
                                           hipStream_t);                             \
  template void launch_ls_scatter_add<T>(T*, const long*, const T*, const T*,        \
                                         const T*, const float*, long, int,          \
                                         hipStream_t);                               \
  template void launch_ls_scatter_bwd<T>(const T*, const long*, const T*, const T*,  \
                                         const T*, const float*, T*, float*, float*, \
                                         long, int, hipStream_t);                    \
  template void launch_row_gather<T>(const T*, const long*, T*, long, int,           \
                                     hipStream_t);                                   \
  template void launch_row_scatter_add<T>(T*, const long*, const T*, const float*,   \
                                          long, int, hipStream_t);                   \
  template void launch_row_gather_scaled<T>(const T*, const long*, const float*, T*, \
                                            long, int, hipStream_t);                 \
  template void launch_bias_gelu_bwd<T>(const T*, const T*, const T*, T*, float*,    \
                                        long, int, hipStream_t);                     \
  template void launch_swiglu_fwd<T>(const T*, T*, long, int, hipStream_t);          \
  template void launch_swiglu_bwd<T>(const T*, const T*, T*, long, int, hipStream_t);\
  template void launch_rope_fwd<T>(const T*, const float*, const float*, T*, long,   \
                                   int, int, int, hipStream_t);

INSTANTIATE_EW(float)
INSTANTIATE_EW(__hip_bfloat16)
