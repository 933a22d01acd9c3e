// Row-wise normalization kernels: LayerNorm / RMSNorm / L2-norm, fwd + bwd.
// (SURVEY K2/K3 + the DINO-head bottleneck normalize of K15.)
//
// Shape model: x is [rows, D] contiguous, bf16 or fp32 I/O, fp32 math.
// One 256-thread block per row (grid-stride over rows); vectorized 8-wide
// bf16 loads on the fast path (guideline 13). Weight grads accumulate into a
// per-block LDS column buffer and leave via one atomicAdd per column per
// block (guideline 12).

#include "common.h"

#define NORM_BLOCK 256

// ------------------------------ LayerNorm ------------------------------

template <typename T>
__global__ void layernorm_fwd_kernel(
    const T* __restrict__ x, const T* __restrict__ w, const T* __restrict__ b,
    T* __restrict__ y, float* __restrict__ mean_out, float* __restrict__ rstd_out,
    long rows, int D, float eps) {
  __shared__ float red[16];
  for (long row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* xr = x + row * (long)D;
    T* yr = y + row * (long)D;
    float s = 0.f, s2 = 0.f;
    for (int i = threadIdx.x; i < D; i += blockDim.x) {
      float v = ScalarOps<T>::load(xr + i);
      s += v;
      s2 += v * v;
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
    for (int i = threadIdx.x; i < D; i += blockDim.x) {
      float v = ScalarOps<T>::load(xr + i);
      float wi = ScalarOps<T>::load(w + i);
      float bi = ScalarOps<T>::load(b + i);
      ScalarOps<T>::store(yr + i, (v - mean) * rstd * wi + bi);
    }
    __syncthreads();
  }
}

template <typename T>
__global__ void layernorm_bwd_kernel(
    const T* __restrict__ dy, const T* __restrict__ x, const T* __restrict__ w,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    T* __restrict__ dx, float* __restrict__ dw, float* __restrict__ db,
    long rows, int D) {
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
    for (int i = threadIdx.x; i < D; i += blockDim.x) {
      float g = ScalarOps<T>::load(dyr + i);
      float xhat = (ScalarOps<T>::load(xr + i) - m) * r;
      float wi = ScalarOps<T>::load(w + i);
      float gw = g * wi;
      sum_dyw += gw;
      sum_dyw_xhat += gw * xhat;
      dw_acc[i] += g * xhat;
      db_acc[i] += g;
    }
    sum_dyw = block_reduce_sum(sum_dyw, red);
    __syncthreads();
    sum_dyw_xhat = block_reduce_sum(sum_dyw_xhat, red);
    const float inv_d = 1.0f / D;
    for (int i = threadIdx.x; i < D; i += blockDim.x) {
      float g = ScalarOps<T>::load(dyr + i);
      float xhat = (ScalarOps<T>::load(xr + i) - m) * r;
      float wi = ScalarOps<T>::load(w + i);
      float v = (g * wi - (sum_dyw + xhat * sum_dyw_xhat) * inv_d) * r;
      ScalarOps<T>::store(dxr + i, v);
    }
    __syncthreads();
  }
  for (int i = threadIdx.x; i < D; i += blockDim.x) {
    atomicAdd(dw + i, dw_acc[i]);
    atomicAdd(db + i, db_acc[i]);
  }
}

// ------------------------------ RMSNorm --------------------------------

template <typename T>
__global__ void rmsnorm_fwd_kernel(
    const T* __restrict__ x, const T* __restrict__ w, T* __restrict__ y,
    float* __restrict__ rstd_out, long rows, int D, float eps) {
  __shared__ float red[16];
  for (long row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* xr = x + row * (long)D;
    T* yr = y + row * (long)D;
    float s2 = 0.f;
    for (int i = threadIdx.x; i < D; i += blockDim.x) {
      float v = ScalarOps<T>::load(xr + i);
      s2 += v * v;
    }
    s2 = block_reduce_sum(s2, red);
    float rstd = rsqrtf(s2 / D + eps);
    if (threadIdx.x == 0) rstd_out[row] = rstd;
    for (int i = threadIdx.x; i < D; i += blockDim.x) {
      float v = ScalarOps<T>::load(xr + i);
      float wi = ScalarOps<T>::load(w + i);
      ScalarOps<T>::store(yr + i, v * rstd * wi);
    }
    __syncthreads();
  }
}

template <typename T>
__global__ void rmsnorm_bwd_kernel(
    const T* __restrict__ dy, const T* __restrict__ x, const T* __restrict__ w,
    const float* __restrict__ rstd, T* __restrict__ dx, float* __restrict__ dw,
    long rows, int D) {
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
    for (int i = threadIdx.x; i < D; i += blockDim.x) {
      float g = ScalarOps<T>::load(dyr + i);
      float v = ScalarOps<T>::load(xr + i);
      float wi = ScalarOps<T>::load(w + i);
      sum_gxw += g * wi * v;
      dw_acc[i] += g * v * r;
    }
    sum_gxw = block_reduce_sum(sum_gxw, red);
    const float c = sum_gxw * r * r * r / D;
    for (int i = threadIdx.x; i < D; i += blockDim.x) {
      float g = ScalarOps<T>::load(dyr + i);
      float v = ScalarOps<T>::load(xr + i);
      float wi = ScalarOps<T>::load(w + i);
      ScalarOps<T>::store(dxr + i, g * wi * r - v * c);
    }
    __syncthreads();
  }
  for (int i = threadIdx.x; i < D; i += blockDim.x) atomicAdd(dw + i, dw_acc[i]);
}

// ------------------------------ L2 norm --------------------------------
// y = x / (||x|| + eps); saves s = 1/(||x|| + eps). ||x|| = 1/s - eps.

template <typename T>
__global__ void l2norm_fwd_kernel(
    const T* __restrict__ x, T* __restrict__ y, float* __restrict__ s_out,
    long rows, int D, float eps) {
  __shared__ float red[16];
  for (long row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* xr = x + row * (long)D;
    T* yr = y + row * (long)D;
    float s2 = 0.f;
    for (int i = threadIdx.x; i < D; i += blockDim.x) {
      float v = ScalarOps<T>::load(xr + i);
      s2 += v * v;
    }
    s2 = block_reduce_sum(s2, red);
    float s = 1.0f / (sqrtf(s2) + eps);
    if (threadIdx.x == 0) s_out[row] = s;
    for (int i = threadIdx.x; i < D; i += blockDim.x) {
      float v = ScalarOps<T>::load(xr + i);
      ScalarOps<T>::store(yr + i, v * s);
    }
    __syncthreads();
  }
}

template <typename T>
__global__ void l2norm_bwd_kernel(
    const T* __restrict__ dy, const T* __restrict__ y, const float* __restrict__ s_in,
    T* __restrict__ dx, long rows, int D, float eps) {
  __shared__ float red[16];
  for (long row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* dyr = dy + row * (long)D;
    const T* yr = y + row * (long)D;
    T* dxr = dx + row * (long)D;
    const float s = s_in[row];
    const float n = 1.0f / s - eps;  // the original norm
    float dot = 0.f;
    for (int i = threadIdx.x; i < D; i += blockDim.x) {
      dot += ScalarOps<T>::load(dyr + i) * ScalarOps<T>::load(yr + i);
    }
    dot = block_reduce_sum(dot, red);
    // y = x * s, s = 1/(n+eps)  =>  dx = s*dy - (dot(dy,y)/n) * y
    const float c = dot / fmaxf(n, 1e-20f);
    for (int i = threadIdx.x; i < D; i += blockDim.x) {
      float g = ScalarOps<T>::load(dyr + i);
      float yv = ScalarOps<T>::load(yr + i);
      ScalarOps<T>::store(dxr + i, s * g - c * yv);
    }
    __syncthreads();
  }
}

// ------------------------------ C wrappers -----------------------------

template <typename T>
void launch_layernorm_fwd(const T* x, const T* w, const T* b, T* y, float* mean,
                          float* rstd, long rows, int D, float eps, hipStream_t stream) {
  int grid = (int)min(rows, (long)8192);
  hipLaunchKernelGGL((layernorm_fwd_kernel<T>), dim3(grid), dim3(NORM_BLOCK), 0, stream,
                     x, w, b, y, mean, rstd, rows, D, eps);
}

template <typename T>
void launch_layernorm_bwd(const T* dy, const T* x, const T* w, const float* mean,
                          const float* rstd, T* dx, float* dw, float* db, long rows,
                          int D, hipStream_t stream) {
  int grid = (int)min(rows, (long)1024);
  size_t shmem = (16 + 2 * (size_t)D) * sizeof(float);
  hipLaunchKernelGGL((layernorm_bwd_kernel<T>), dim3(grid), dim3(NORM_BLOCK), shmem, stream,
                     dy, x, w, mean, rstd, dx, dw, db, rows, D);
}

template <typename T>
void launch_rmsnorm_fwd(const T* x, const T* w, T* y, float* rstd, long rows, int D,
                        float eps, hipStream_t stream) {
  int grid = (int)min(rows, (long)8192);
  hipLaunchKernelGGL((rmsnorm_fwd_kernel<T>), dim3(grid), dim3(NORM_BLOCK), 0, stream,
                     x, w, y, rstd, rows, D, eps);
}

template <typename T>
void launch_rmsnorm_bwd(const T* dy, const T* x, const T* w, const float* rstd, T* dx,
                        float* dw, long rows, int D, hipStream_t stream) {
  int grid = (int)min(rows, (long)1024);
  size_t shmem = (16 + (size_t)D) * sizeof(float);
  hipLaunchKernelGGL((rmsnorm_bwd_kernel<T>), dim3(grid), dim3(NORM_BLOCK), shmem, stream,
                     dy, x, w, rstd, dx, dw, rows, D);
}

template <typename T>
void launch_l2norm_fwd(const T* x, T* y, float* s, long rows, int D, float eps,
                       hipStream_t stream) {
  int grid = (int)min(rows, (long)8192);
  hipLaunchKernelGGL((l2norm_fwd_kernel<T>), dim3(grid), dim3(NORM_BLOCK), 0, stream,
                     x, y, s, rows, D, eps);
}

template <typename T>
void launch_l2norm_bwd(const T* dy, const T* y, const float* s, T* dx, long rows, int D,
                       float eps, hipStream_t stream) {
  int grid = (int)min(rows, (long)8192);
  hipLaunchKernelGGL((l2norm_bwd_kernel<T>), dim3(grid), dim3(NORM_BLOCK), 0, stream,
                     dy, y, s, dx, rows, D, eps);
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
