// Row-wise normalization kernels: LayerNorm / RMSNorm / L2-norm, fwd + bwd.
// (SURVEY K2/K3 + the DINO-head bottleneck normalize of K15.)
//
// x is [rows, D] contiguous, bf16 or fp32 I/O, fp32 math. One 256-thread
// block per row (grid-stride over rows). V8=true takes the 8-wide vector
// path (16 B/lane bf16 loads — guideline 13); weight grads accumulate in
// registers per fixed column slice in LDS and leave via one atomicAdd per
// column per block.

#include "common.h"

// power of two: number of dgamma/dbeta shadow accumulators (see layernorm_bwd)
#define LN_SHADOWS 32

#define NORM_BLOCK 256

// block size matched to the row width so no lanes idle in the 8-wide loops
static inline int norm_block_for(int D) {
  if (D >= 2048) return 256;
  if (D >= 1024) return 128;
  return 64;
}

template <typename T, bool V8>
DEV_INLINE void row_load8(const T* p, int i, float* out) {
  if constexpr (V8) {
    T buf[8];
    Vec8<T>::load(buf, p + i);
#pragma unroll
    for (int e = 0; e < 8; ++e) out[e] = ScalarOps<T>::load(buf + e);
  } else {
    out[0] = ScalarOps<T>::load(p + i);
  }
}

template <typename T, bool V8>
DEV_INLINE void row_store8(T* p, int i, const float* in) {
  if constexpr (V8) {
    T buf[8];
#pragma unroll
    for (int e = 0; e < 8; ++e) ScalarOps<T>::store(buf + e, in[e]);
    Vec8<T>::store(p + i, buf);
  } else {
    ScalarOps<T>::store(p + i, in[0]);
  }
}

#define ROW_LOOP(i) \
  for (int i = threadIdx.x * EW; i < D; i += blockDim.x * EW)

// ------------------------------ LayerNorm ------------------------------

template <typename T, bool V8>
__global__ void layernorm_fwd_kernel(
    const T* __restrict__ x, const T* __restrict__ w, const T* __restrict__ b,
    T* __restrict__ y, float* __restrict__ mean_out, float* __restrict__ rstd_out,
    long rows, int D, float eps) {
  constexpr int EW = V8 ? 8 : 1;
  __shared__ float red[16];
  for (long row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* xr = x + row * (long)D;
    T* yr = y + row * (long)D;
    float s = 0.f, s2 = 0.f;
    ROW_LOOP(i) {
      float v[EW];
      row_load8<T, V8>(xr, i, v);
#pragma unroll
      for (int e = 0; e < EW; ++e) {
        s += v[e];
        s2 += v[e] * v[e];
      }
    }
    s = block_reduce_sum(s, red);
    __syncthreads();
    s2 = block_reduce_sum(s2, red);
    float mean = s / D;
    float var = s2 / D - mean * mean;
    float rstd = rsqrtf(fmaxf(var, 0.f) + eps);
    if (threadIdx.x == 0) {
      mean_out[row] = mean;
      rstd_out[row] = rstd;
    }
    ROW_LOOP(i) {
      float v[EW], wv[EW], bv[EW], o[EW];
      row_load8<T, V8>(xr, i, v);
      row_load8<T, V8>(w, i, wv);
      row_load8<T, V8>(b, i, bv);
#pragma unroll
      for (int e = 0; e < EW; ++e) o[e] = (v[e] - mean) * rstd * wv[e] + bv[e];
      row_store8<T, V8>(yr, i, o);
    }
    __syncthreads();
  }
}

template <typename T, bool V8>
__global__ void layernorm_bwd_kernel(
    const T* __restrict__ dy, const T* __restrict__ x, const T* __restrict__ w,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    T* __restrict__ dx, float* __restrict__ dw, float* __restrict__ db,
    long rows, int D) {
  constexpr int EW = V8 ? 8 : 1;
  extern __shared__ float smem[];  // [16 red] + [D dw] + [D db]
  float* red = smem;
  float* dw_acc = smem + 16;
  float* db_acc = dw_acc + D;
  for (int i = threadIdx.x; i < D; i += blockDim.x) {
    dw_acc[i] = 0.f;
    db_acc[i] = 0.f;
  }
  __syncthreads();
  for (long row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* dyr = dy + row * (long)D;
    const T* xr = x + row * (long)D;
    T* dxr = dx + row * (long)D;
    const float m = mean[row], r = rstd[row];
    float sum_dyw = 0.f, sum_dyw_xhat = 0.f;
    ROW_LOOP(i) {
      float g[EW], v[EW], wv[EW];
      row_load8<T, V8>(dyr, i, g);
      row_load8<T, V8>(xr, i, v);
      row_load8<T, V8>(w, i, wv);
#pragma unroll
      for (int e = 0; e < EW; ++e) {
        const float xhat = (v[e] - m) * r;
        const float gw = g[e] * wv[e];
        sum_dyw += gw;
        sum_dyw_xhat += gw * xhat;
        dw_acc[i + e] += g[e] * xhat;
        db_acc[i + e] += g[e];
      }
    }
    sum_dyw = block_reduce_sum(sum_dyw, red);
    __syncthreads();
    sum_dyw_xhat = block_reduce_sum(sum_dyw_xhat, red);
    const float inv_d = 1.0f / D;
    ROW_LOOP(i) {
      float g[EW], v[EW], wv[EW], o[EW];
      row_load8<T, V8>(dyr, i, g);
      row_load8<T, V8>(xr, i, v);
      row_load8<T, V8>(w, i, wv);
#pragma unroll
      for (int e = 0; e < EW; ++e) {
        const float xhat = (v[e] - m) * r;
        o[e] = (g[e] * wv[e] - (sum_dyw + xhat * sum_dyw_xhat) * inv_d) * r;
      }
      row_store8<T, V8>(dxr, i, o);
    }
    __syncthreads();
  }
  // shadow copies: chain depth of the per-address atomic RMW is
  // gridDim/shadows instead of gridDim (dw/db are [shadows, D], summed by
  // the caller)
  float* dw_s = dw + (long)(blockIdx.x & (LN_SHADOWS - 1)) * D;
  float* db_s = db + (long)(blockIdx.x & (LN_SHADOWS - 1)) * D;
  for (int i = threadIdx.x; i < D; i += blockDim.x) {
    atomicAdd(dw_s + i, dw_acc[i]);
    atomicAdd(db_s + i, db_acc[i]);
  }
}

// Wave-per-row LayerNorm backward: each 64-lane wave owns a row, the row's
// dy/x/w values stay register-cached between the reduction and the dx pass
// (ONE read of dy and x instead of two), and the row loop has no block
// barriers — wave-level __shfl_xor reductions only. Per-wave dgamma/dbeta
// partials live in LDS and merge once at block end into the shadowed global
// accumulators. CHUNKS = ceil(D/512) (1: D<=512, 2: 1024, 3: 1536).
template <typename T, int CHUNKS>
__global__ __launch_bounds__(256) void layernorm_bwd_wave_kernel(
    const T* __restrict__ dy, const T* __restrict__ x, const T* __restrict__ w,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    T* __restrict__ dx, float* __restrict__ dw, float* __restrict__ db,
    long rows, int D) {
  constexpr int NW = 4;  // waves per 256-thread block
  const int wid = threadIdx.x / 64;
  const int lane = threadIdx.x & 63;
  extern __shared__ float smem[];  // [NW][D] dw partials + [NW][D] db
  float* dw_acc = smem + (long)wid * D;
  float* db_acc = smem + (long)(NW + wid) * D;
  for (int i = lane; i < D; i += 64) {
    dw_acc[i] = 0.f;
    db_acc[i] = 0.f;
  }
  float g[CHUNKS][8], v[CHUNKS][8], wv[CHUNKS][8];
  for (long row = (long)blockIdx.x * NW + wid; row < rows; row += (long)gridDim.x * NW) {
    const T* dyr = dy + row * (long)D;
    const T* xr = x + row * (long)D;
    const float m = mean[row], r = rstd[row];
    float sum_dyw = 0.f, sum_dyw_xhat = 0.f;
#pragma unroll
    for (int c = 0; c < CHUNKS; ++c) {
      const int i = (c * 64 + lane) * 8;
      if (i < D) {
        row_load8<T, true>(dyr, i, g[c]);
        row_load8<T, true>(xr, i, v[c]);
        row_load8<T, true>(w, i, wv[c]);
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          const float xhat = (v[c][e] - m) * r;
          const float gw = g[c][e] * wv[c][e];
          sum_dyw += gw;
          sum_dyw_xhat += gw * xhat;
          dw_acc[i + e] += g[c][e] * xhat;
          db_acc[i + e] += g[c][e];
        }
      }
    }
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      sum_dyw += __shfl_xor(sum_dyw, off, 64);
      sum_dyw_xhat += __shfl_xor(sum_dyw_xhat, off, 64);
    }
    const float inv_d = 1.0f / D;
    T* dxr = dx + row * (long)D;
#pragma unroll
    for (int c = 0; c < CHUNKS; ++c) {
      const int i = (c * 64 + lane) * 8;
      if (i < D) {
        float o[8];
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          const float xhat = (v[c][e] - m) * r;
          o[e] = (g[c][e] * wv[c][e] - (sum_dyw + xhat * sum_dyw_xhat) * inv_d) * r;
        }
        row_store8<T, true>(dxr, i, o);
      }
    }
  }
  __syncthreads();
  const long sh = (long)(blockIdx.x & (LN_SHADOWS - 1)) * D;
  for (int i = threadIdx.x; i < D; i += 256) {
    float a = 0.f, bsum = 0.f;
#pragma unroll
    for (int w2 = 0; w2 < NW; ++w2) {
      a += smem[(long)w2 * D + i];
      bsum += smem[(long)(NW + w2) * D + i];
    }
    atomicAdd(dw + sh + i, a);
    atomicAdd(db + sh + i, bsum);
  }
}

// ------------------------------ RMSNorm --------------------------------

template <typename T, bool V8>
__global__ void rmsnorm_fwd_kernel(
    const T* __restrict__ x, const T* __restrict__ w, T* __restrict__ y,
    float* __restrict__ rstd_out, long rows, int D, float eps) {
  constexpr int EW = V8 ? 8 : 1;
  __shared__ float red[16];
  for (long row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* xr = x + row * (long)D;
    T* yr = y + row * (long)D;
    float s2 = 0.f;
    ROW_LOOP(i) {
      float v[EW];
      row_load8<T, V8>(xr, i, v);
#pragma unroll
      for (int e = 0; e < EW; ++e) s2 += v[e] * v[e];
    }
    s2 = block_reduce_sum(s2, red);
    float rstd = rsqrtf(s2 / D + eps);
    if (threadIdx.x == 0) rstd_out[row] = rstd;
    ROW_LOOP(i) {
      float v[EW], wv[EW], o[EW];
      row_load8<T, V8>(xr, i, v);
      row_load8<T, V8>(w, i, wv);
#pragma unroll
      for (int e = 0; e < EW; ++e) o[e] = v[e] * rstd * wv[e];
      row_store8<T, V8>(yr, i, o);
    }
    __syncthreads();
  }
}

template <typename T, bool V8>
__global__ void rmsnorm_bwd_kernel(
    const T* __restrict__ dy, const T* __restrict__ x, const T* __restrict__ w,
    const float* __restrict__ rstd, T* __restrict__ dx, float* __restrict__ dw,
    long rows, int D) {
  constexpr int EW = V8 ? 8 : 1;
  extern __shared__ float smem[];
  float* red = smem;
  float* dw_acc = smem + 16;
  for (int i = threadIdx.x; i < D; i += blockDim.x) dw_acc[i] = 0.f;
  __syncthreads();
  for (long row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* dyr = dy + row * (long)D;
    const T* xr = x + row * (long)D;
    T* dxr = dx + row * (long)D;
    const float r = rstd[row];
    float sum_gxw = 0.f;
    ROW_LOOP(i) {
      float g[EW], v[EW], wv[EW];
      row_load8<T, V8>(dyr, i, g);
      row_load8<T, V8>(xr, i, v);
      row_load8<T, V8>(w, i, wv);
#pragma unroll
      for (int e = 0; e < EW; ++e) {
        sum_gxw += g[e] * wv[e] * v[e];
        dw_acc[i + e] += g[e] * v[e] * r;
      }
    }
    sum_gxw = block_reduce_sum(sum_gxw, red);
    const float c = sum_gxw * r * r * r / D;
    ROW_LOOP(i) {
      float g[EW], v[EW], wv[EW], o[EW];
      row_load8<T, V8>(dyr, i, g);
      row_load8<T, V8>(xr, i, v);
      row_load8<T, V8>(w, i, wv);
#pragma unroll
      for (int e = 0; e < EW; ++e) o[e] = g[e] * wv[e] * r - v[e] * c;
      row_store8<T, V8>(dxr, i, o);
    }
    __syncthreads();
  }
  for (int i = threadIdx.x; i < D; i += blockDim.x) atomicAdd(dw + i, dw_acc[i]);
}

// ------------------------------ L2 norm --------------------------------
// y = x / (||x|| + eps); saves s = 1/(||x|| + eps). ||x|| = 1/s - eps.

template <typename T, bool V8>
__global__ void l2norm_fwd_kernel(
    const T* __restrict__ x, T* __restrict__ y, float* __restrict__ s_out,
    long rows, int D, float eps) {
  constexpr int EW = V8 ? 8 : 1;
  __shared__ float red[16];
  for (long row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* xr = x + row * (long)D;
    T* yr = y + row * (long)D;
    float s2 = 0.f;
    ROW_LOOP(i) {
      float v[EW];
      row_load8<T, V8>(xr, i, v);
#pragma unroll
      for (int e = 0; e < EW; ++e) s2 += v[e] * v[e];
    }
    s2 = block_reduce_sum(s2, red);
    float s = 1.0f / (sqrtf(s2) + eps);
    if (threadIdx.x == 0) s_out[row] = s;
    ROW_LOOP(i) {
      float v[EW], o[EW];
      row_load8<T, V8>(xr, i, v);
#pragma unroll
      for (int e = 0; e < EW; ++e) o[e] = v[e] * s;
      row_store8<T, V8>(yr, i, o);
    }
    __syncthreads();
  }
}

template <typename T, bool V8>
__global__ void l2norm_bwd_kernel(
    const T* __restrict__ dy, const T* __restrict__ y, const float* __restrict__ s_in,
    T* __restrict__ dx, long rows, int D, float eps) {
  constexpr int EW = V8 ? 8 : 1;
  __shared__ float red[16];
  for (long row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* dyr = dy + row * (long)D;
    const T* yr = y + row * (long)D;
    T* dxr = dx + row * (long)D;
    const float s = s_in[row];
    const float n = 1.0f / s - eps;
    float dot = 0.f;
    ROW_LOOP(i) {
      float g[EW], v[EW];
      row_load8<T, V8>(dyr, i, g);
      row_load8<T, V8>(yr, i, v);
#pragma unroll
      for (int e = 0; e < EW; ++e) dot += g[e] * v[e];
    }
    dot = block_reduce_sum(dot, red);
    // y = x * s, s = 1/(n+eps)  =>  dx = s*dy - (dot(dy,y)/n) * y
    const float c = dot / fmaxf(n, 1e-20f);
    ROW_LOOP(i) {
      float g[EW], v[EW], o[EW];
      row_load8<T, V8>(dyr, i, g);
      row_load8<T, V8>(yr, i, v);
#pragma unroll
      for (int e = 0; e < EW; ++e) o[e] = s * g[e] - c * v[e];
      row_store8<T, V8>(dxr, i, o);
    }
    __syncthreads();
  }
}

// ------------------------------ C wrappers -----------------------------

#define DISPATCH_V8(D, ...)            \
  if ((D) % 8 == 0) {                  \
    constexpr bool V8 = true;          \
    __VA_ARGS__;                       \
  } else {                             \
    constexpr bool V8 = false;         \
    __VA_ARGS__;                       \
  }

template <typename T>
void launch_layernorm_fwd(const T* x, const T* w, const T* b, T* y, float* mean,
                          float* rstd, long rows, int D, float eps, hipStream_t stream) {
  int grid = (int)min(rows, (long)8192);
  DISPATCH_V8(D, hipLaunchKernelGGL(HIP_KERNEL_NAME(layernorm_fwd_kernel<T, V8>), dim3(grid),
                                     dim3(norm_block_for(D)), 0, stream, x, w, b, y, mean, rstd,
                                     rows, D, eps));
}

template <typename T>
void launch_layernorm_bwd(const T* dy, const T* x, const T* w, const float* mean,
                          const float* rstd, T* dx, float* dw, float* db, long rows,
                          int D, hipStream_t stream) {
  if (D % 8 == 0 && D <= 1536) {
    // wave-per-row: one read of dy/x, no block barriers in the row loop
    int grid = (int)min((rows + 3) / 4, (long)2048);
    size_t shmem = 2 * 4 * (size_t)D * sizeof(float);
    if (D <= 512)
      hipLaunchKernelGGL(HIP_KERNEL_NAME(layernorm_bwd_wave_kernel<T, 1>), dim3(grid),
                         dim3(256), shmem, stream, dy, x, w, mean, rstd, dx, dw, db,
                         rows, D);
    else if (D <= 1024)
      hipLaunchKernelGGL(HIP_KERNEL_NAME(layernorm_bwd_wave_kernel<T, 2>), dim3(grid),
                         dim3(256), shmem, stream, dy, x, w, mean, rstd, dx, dw, db,
                         rows, D);
    else
      hipLaunchKernelGGL(HIP_KERNEL_NAME(layernorm_bwd_wave_kernel<T, 3>), dim3(grid),
                         dim3(256), shmem, stream, dy, x, w, mean, rstd, dx, dw, db,
                         rows, D);
    return;
  }
  int grid = (int)min(rows, (long)2048);
  size_t shmem = (16 + 2 * (size_t)D) * sizeof(float);
  DISPATCH_V8(D, hipLaunchKernelGGL(HIP_KERNEL_NAME(layernorm_bwd_kernel<T, V8>), dim3(grid),
                                     dim3(norm_block_for(D)), shmem, stream, dy, x, w, mean,
                                     rstd, dx, dw, db, rows, D));
}

template <typename T>
void launch_rmsnorm_fwd(const T* x, const T* w, T* y, float* rstd, long rows, int D,
                        float eps, hipStream_t stream) {
  int grid = (int)min(rows, (long)8192);
  DISPATCH_V8(D, hipLaunchKernelGGL(HIP_KERNEL_NAME(rmsnorm_fwd_kernel<T, V8>), dim3(grid),
                                     dim3(norm_block_for(D)), 0, stream, x, w, y, rstd, rows, D,
                                     eps));
}

template <typename T>
void launch_rmsnorm_bwd(const T* dy, const T* x, const T* w, const float* rstd, T* dx,
                        float* dw, long rows, int D, hipStream_t stream) {
  int grid = (int)min(rows, (long)2048);
  size_t shmem = (16 + (size_t)D) * sizeof(float);
  DISPATCH_V8(D, hipLaunchKernelGGL(HIP_KERNEL_NAME(rmsnorm_bwd_kernel<T, V8>), dim3(grid),
                                     dim3(norm_block_for(D)), shmem, stream, dy, x, w, rstd, dx,
                                     dw, rows, D));
}

template <typename T>
void launch_l2norm_fwd(const T* x, T* y, float* s, long rows, int D, float eps,
                       hipStream_t stream) {
  int grid = (int)min(rows, (long)8192);
  DISPATCH_V8(D, hipLaunchKernelGGL(HIP_KERNEL_NAME(l2norm_fwd_kernel<T, V8>), dim3(grid),
                                     dim3(norm_block_for(D)), 0, stream, x, y, s, rows, D, eps));
}

template <typename T>
void launch_l2norm_bwd(const T* dy, const T* y, const float* s, T* dx, long rows, int D,
                       float eps, hipStream_t stream) {
  int grid = (int)min(rows, (long)8192);
  DISPATCH_V8(D, hipLaunchKernelGGL(HIP_KERNEL_NAME(l2norm_bwd_kernel<T, V8>), dim3(grid),
                                     dim3(norm_block_for(D)), 0, stream, dy, y, s, dx, rows, D,
                                     eps));
}

// explicit instantiations used by bindings.cpp
#define INSTANTIATE_NORMS(T)                                                              \
  template void launch_layernorm_fwd<T>(const T*, const T*, const T*, T*, float*, float*, \
                                        long, int, float, hipStream_t);                   \
  template void launch_layernorm_bwd<T>(const T*, const T*, const T*, const float*,       \
                                        const float*, T*, float*, float*, long, int,      \
                                        hipStream_t);                                     \
  template void launch_rmsnorm_fwd<T>(const T*, const T*, T*, float*, long, int, float,   \
                                      hipStream_t);                                       \
  template void launch_rmsnorm_bwd<T>(const T*, const T*, const T*, const float*, T*,     \
                                      float*, long, int, hipStream_t);                    \
  template void launch_l2norm_fwd<T>(const T*, T*, float*, long, int, float, hipStream_t);\
  template void launch_l2norm_bwd<T>(const T*, const T*, const float*, T*, long, int,     \
                                     float, hipStream_t);

INSTANTIATE_NORMS(float)
INSTANTIATE_NORMS(__hip_bfloat16)
