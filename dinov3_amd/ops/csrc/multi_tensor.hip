// Multi-tensor fused update kernels (SURVEY K23 EMA / K24 AdamW / K25 norm).
//
// One launch walks a chunk table covering every tensor in the list: chunk c
// belongs to tensor tensor_id[c] at element offset chunk_off[c]. The tables
// live in a small device buffer uploaded once per call (bindings.cpp caches
// the flattened layout per tensor-list where profitable).

#include "common.h"

#define MT_BLOCK 256
#define MT_CHUNK 65536  // elements per chunk per block

struct ChunkTable {
  const long* tensor_sizes;   // [n_tensors]
  const int* chunk_tensor;    // [n_chunks] -> tensor id
  const long* chunk_offset;   // [n_chunks] -> start element within tensor
};

// ------------------------------- EMA -----------------------------------
// t = m*t + (1-m)*s, in place; teacher bf16 (or fp32), student same dtype.

template <typename T>
__global__ void multi_tensor_ema_kernel(
    T* const* __restrict__ t_ptrs, const T* const* __restrict__ s_ptrs,
    const long* __restrict__ sizes, const int* __restrict__ chunk_tensor,
    const long* __restrict__ chunk_offset, int n_chunks, float m) {
  for (int c = blockIdx.x; c < n_chunks; c += gridDim.x) {
    const int tid = chunk_tensor[c];
    const long off = chunk_offset[c];
    const long n = min((long)MT_CHUNK, sizes[tid] - off);
    T* t = t_ptrs[tid] + off;
    const T* s = s_ptrs[tid] + off;
    for (long i = threadIdx.x; i < n; i += blockDim.x) {
      float tv = ScalarOps<T>::load(t + i);
      float sv = ScalarOps<T>::load(s + i);
      ScalarOps<T>::store(t + i, m * tv + (1.0f - m) * sv);
    }
  }
}

// ------------------------------ AdamW ----------------------------------
// Params may be bf16 with fp32 master weights (master != nullptr): the update
// runs on master and re-quantizes params. exp_avg/exp_avg_sq are fp32.

template <typename T, bool HAS_MASTER>
__global__ void multi_tensor_adamw_kernel(
    T* const* __restrict__ p_ptrs, const T* const* __restrict__ g_ptrs,
    float* const* __restrict__ m_ptrs, float* const* __restrict__ v_ptrs,
    float* const* __restrict__ w_ptrs,  // masters (nullable)
    const long* __restrict__ sizes, const int* __restrict__ chunk_tensor,
    const long* __restrict__ chunk_offset, int n_chunks,
    float lr, float beta1, float beta2, float eps, float weight_decay,
    float bc1, float bc2, float grad_scale) {
  for (int c = blockIdx.x; c < n_chunks; c += gridDim.x) {
    const int tid = chunk_tensor[c];
    const long off = chunk_offset[c];
    const long n = min((long)MT_CHUNK, sizes[tid] - off);
    T* p = p_ptrs[tid] + off;
    const T* g = g_ptrs[tid] + off;
    float* m = m_ptrs[tid] + off;
    float* v = v_ptrs[tid] + off;
    float* w = HAS_MASTER ? (w_ptrs[tid] + off) : nullptr;
    const float inv_bc1 = 1.0f / bc1;
    const float inv_bc2 = 1.0f / bc2;
    for (long i = threadIdx.x; i < n; i += blockDim.x) {
      float gv = ScalarOps<T>::load(g + i) * grad_scale;
      float mv = beta1 * m[i] + (1.0f - beta1) * gv;
      float vv = beta2 * v[i] + (1.0f - beta2) * gv * gv;
      m[i] = mv;
      v[i] = vv;
      float pv = HAS_MASTER ? w[i] : ScalarOps<T>::load(p + i);
      pv *= (1.0f - lr * weight_decay);
      pv -= lr * (mv * inv_bc1) / (sqrtf(vv * inv_bc2) + eps);
      if (HAS_MASTER) w[i] = pv;
      ScalarOps<T>::store(p + i, pv);
    }
  }
}

// ----------------------------- L2 norm^2 --------------------------------

template <typename T>
__global__ void multi_tensor_l2norm_sq_kernel(
    const T* const* __restrict__ g_ptrs, const long* __restrict__ sizes,
    const int* __restrict__ chunk_tensor, const long* __restrict__ chunk_offset,
    int n_chunks, float* __restrict__ out) {
  __shared__ float red[16];
  float acc = 0.f;
  for (int c = blockIdx.x; c < n_chunks; c += gridDim.x) {
    const int tid = chunk_tensor[c];
    const long off = chunk_offset[c];
    const long n = min((long)MT_CHUNK, sizes[tid] - off);
    const T* g = g_ptrs[tid] + off;
    for (long i = threadIdx.x; i < n; i += blockDim.x) {
      float v = ScalarOps<T>::load(g + i);
      acc += v * v;
    }
  }
  acc = block_reduce_sum(acc, red);
  if (threadIdx.x == 0) atomicAdd(out, acc);
}

// ----------------------- planned (single-launch) ------------------------
// Per-tensor hyperparameter arrays let ALL fused param groups run in one
// launch: lr_eff = (is_last ? last_lr : lr) * lr_mult[t]; wd_eff = wd *
// wd_mult[t]; grad scale = clip[submodel_id[t]] (per-submodel global-norm
// clip factors, uploaded per step as a tiny array).

template <typename T, typename GT, bool HAS_MASTER>
__global__ void multi_tensor_adamw_planned_kernel(
    T* const* __restrict__ p_ptrs, const GT* const* __restrict__ g_ptrs,
    float* const* __restrict__ m_ptrs, float* const* __restrict__ v_ptrs,
    float* const* __restrict__ w_ptrs, const long* __restrict__ sizes,
    const int* __restrict__ chunk_tensor, const long* __restrict__ chunk_offset,
    int n_chunks, const float* __restrict__ lr_mult, const float* __restrict__ wd_mult,
    const float* __restrict__ is_last, const int* __restrict__ sub_id,
    const float* __restrict__ clip, float lr, float last_lr, float wd, float beta1,
    float beta2, float eps, float bc1, float bc2) {
  const float inv_bc1 = 1.0f / bc1;
  const float inv_bc2 = 1.0f / bc2;
  for (int c = blockIdx.x; c < n_chunks; c += gridDim.x) {
    const int tid = chunk_tensor[c];
    const long off = chunk_offset[c];
    const long n = min((long)MT_CHUNK, sizes[tid] - off);
    T* p = p_ptrs[tid] + off;
    const GT* g = g_ptrs[tid] + off;
    float* m = m_ptrs[tid] + off;
    float* v = v_ptrs[tid] + off;
    float* w = HAS_MASTER ? (w_ptrs[tid] + off) : nullptr;
    const float glr = (is_last[tid] != 0.f ? last_lr : lr) * lr_mult[tid];
    const float gwd = wd * wd_mult[tid];
    const float gscale = clip[sub_id[tid]];
    for (long i = threadIdx.x; i < n; i += blockDim.x) {
      float gv = ScalarOps<GT>::load(g + i) * gscale;
      float mv = beta1 * m[i] + (1.0f - beta1) * gv;
      float vv = beta2 * v[i] + (1.0f - beta2) * gv * gv;
      m[i] = mv;
      v[i] = vv;
      float pv = HAS_MASTER ? w[i] : ScalarOps<T>::load(p + i);
      pv *= (1.0f - glr * gwd);
      pv -= glr * (mv * inv_bc1) / (sqrtf(vv * inv_bc2) + eps);
      if (HAS_MASTER) w[i] = pv;
      ScalarOps<T>::store(p + i, pv);
    }
  }
}

// grad sum-of-squares per submodel: out[sub_id[t]] += sum g^2
template <typename T>
__global__ void multi_tensor_l2norm_planned_kernel(
    const T* const* __restrict__ g_ptrs, const long* __restrict__ sizes,
    const int* __restrict__ chunk_tensor, const long* __restrict__ chunk_offset,
    int n_chunks, const int* __restrict__ sub_id, float* __restrict__ out) {
  __shared__ float red[16];
  // chunks of one submodel are contiguous in the plan; still reduce per chunk
  for (int c = blockIdx.x; c < n_chunks; c += gridDim.x) {
    const int tid = chunk_tensor[c];
    const long off = chunk_offset[c];
    const long n = min((long)MT_CHUNK, sizes[tid] - off);
    const T* g = g_ptrs[tid] + off;
    float acc = 0.f;
    for (long i = threadIdx.x; i < n; i += blockDim.x) {
      float v = ScalarOps<T>::load(g + i);
      acc += v * v;
    }
    acc = block_reduce_sum(acc, red);
    if (threadIdx.x == 0) atomicAdd(out + sub_id[tid], acc);
    __syncthreads();
  }
}

template <typename T, typename GT>
void launch_multi_tensor_adamw_planned(
    T* const* p, const GT* const* g, float* const* m, float* const* v, float* const* w,
    const long* sizes, const int* chunk_tensor, const long* chunk_offset, int n_chunks,
    const float* lr_mult, const float* wd_mult, const float* is_last, const int* sub_id,
    const float* clip, float lr, float last_lr, float wd, float beta1, float beta2,
    float eps, float bc1, float bc2, bool has_master, hipStream_t stream) {
  int grid = min(n_chunks, 4096);
  if (has_master) {
    hipLaunchKernelGGL(HIP_KERNEL_NAME(multi_tensor_adamw_planned_kernel<T, GT, true>),
                       dim3(grid), dim3(MT_BLOCK), 0, stream, p, g, m, v, w, sizes,
                       chunk_tensor, chunk_offset, n_chunks, lr_mult, wd_mult, is_last,
                       sub_id, clip, lr, last_lr, wd, beta1, beta2, eps, bc1, bc2);
  } else {
    hipLaunchKernelGGL(HIP_KERNEL_NAME(multi_tensor_adamw_planned_kernel<T, GT, false>),
                       dim3(grid), dim3(MT_BLOCK), 0, stream, p, g, m, v, w, sizes,
                       chunk_tensor, chunk_offset, n_chunks, lr_mult, wd_mult, is_last,
                       sub_id, clip, lr, last_lr, wd, beta1, beta2, eps, bc1, bc2);
  }
}

template <typename T>
void launch_multi_tensor_l2norm_planned(const T* const* g, const long* sizes,
                                        const int* chunk_tensor, const long* chunk_offset,
                                        int n_chunks, const int* sub_id, float* out,
                                        hipStream_t stream) {
  int grid = min(n_chunks, 4096);
  hipLaunchKernelGGL((multi_tensor_l2norm_planned_kernel<T>), dim3(grid), dim3(MT_BLOCK),
                     0, stream, g, sizes, chunk_tensor, chunk_offset, n_chunks, sub_id,
                     out);
}

#define INSTANTIATE_MT_PLANNED(T, GT)                                                     \
  template void launch_multi_tensor_adamw_planned<T, GT>(                                 \
      T* const*, const GT* const*, float* const*, float* const*, float* const*,          \
      const long*, const int*, const long*, int, const float*, const float*,             \
      const float*, const int*, const float*, float, float, float, float, float, float,  \
      float, float, bool, hipStream_t);

INSTANTIATE_MT_PLANNED(float, float)
INSTANTIATE_MT_PLANNED(__hip_bfloat16, __hip_bfloat16)
INSTANTIATE_MT_PLANNED(__hip_bfloat16, float)

#define INSTANTIATE_MT_L2P(T)                                                             \
  template void launch_multi_tensor_l2norm_planned<T>(const T* const*, const long*,       \
                                                      const int*, const long*, int,       \
                                                      const int*, float*, hipStream_t);

INSTANTIATE_MT_L2P(float)
INSTANTIATE_MT_L2P(__hip_bfloat16)

// ---------------------------- C wrappers -------------------------------

template <typename T>
void launch_multi_tensor_ema(T* const* t_ptrs, const T* const* s_ptrs, const long* sizes,
                             const int* chunk_tensor, const long* chunk_offset,
                             int n_chunks, float m, hipStream_t stream) {
  int grid = min(n_chunks, 2048);
  hipLaunchKernelGGL((multi_tensor_ema_kernel<T>), dim3(grid), dim3(MT_BLOCK), 0, stream,
                     t_ptrs, s_ptrs, sizes, chunk_tensor, chunk_offset, n_chunks, m);
}

template <typename T>
void launch_multi_tensor_adamw(T* const* p, const T* const* g, float* const* m,
                               float* const* v, float* const* w, const long* sizes,
                               const int* chunk_tensor, const long* chunk_offset,
                               int n_chunks, float lr, float beta1, float beta2,
                               float eps, float weight_decay, float bc1, float bc2,
                               float grad_scale, bool has_master, hipStream_t stream) {
  int grid = min(n_chunks, 2048);
  if (has_master) {
    hipLaunchKernelGGL((multi_tensor_adamw_kernel<T, true>), dim3(grid), dim3(MT_BLOCK), 0,
                       stream, p, g, m, v, w, sizes, chunk_tensor, chunk_offset, n_chunks,
                       lr, beta1, beta2, eps, weight_decay, bc1, bc2, grad_scale);
  } else {
    hipLaunchKernelGGL((multi_tensor_adamw_kernel<T, false>), dim3(grid), dim3(MT_BLOCK), 0,
                       stream, p, g, m, v, w, sizes, chunk_tensor, chunk_offset, n_chunks,
                       lr, beta1, beta2, eps, weight_decay, bc1, bc2, grad_scale);
  }
}

template <typename T>
void launch_multi_tensor_l2norm_sq(const T* const* g, const long* sizes,
                                   const int* chunk_tensor, const long* chunk_offset,
                                   int n_chunks, float* out, hipStream_t stream) {
  int grid = min(n_chunks, 2048);
  hipLaunchKernelGGL((multi_tensor_l2norm_sq_kernel<T>), dim3(grid), dim3(MT_BLOCK), 0,
                     stream, g, sizes, chunk_tensor, chunk_offset, n_chunks, out);
}

#define INSTANTIATE_MT(T)                                                                  \
  template void launch_multi_tensor_ema<T>(T* const*, const T* const*, const long*,        \
                                           const int*, const long*, int, float,            \
                                           hipStream_t);                                   \
  template void launch_multi_tensor_adamw<T>(T* const*, const T* const*, float* const*,    \
                                             float* const*, float* const*, const long*,    \
                                             const int*, const long*, int, float, float,   \
                                             float, float, float, float, float, float,     \
                                             bool, hipStream_t);                           \
  template void launch_multi_tensor_l2norm_sq<T>(const T* const*, const long*, const int*, \
                                                 const long*, int, float*, hipStream_t);

INSTANTIATE_MT(float)
INSTANTIATE_MT(__hip_bfloat16)
