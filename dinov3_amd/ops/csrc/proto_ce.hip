// Fused prototype-score cross-entropy over the K=65536 axis
// and fused Sinkhorn-Knopp iteration kernels (K17).
//
// The DINO loss -sum_{s,t,b,k} t[t,b,k] * log_softmax(x[s,b,k]/temp) never
// materializes the [rows, 65536] softmax: forward is two passes over the
// student row (max, then sum-exp + teacher dot), saving only the row
// logsumexp; backward recomputes softmax from the lse. This replaces a
// 72 ms/step fp32 fallback GEMM + fp32 log_softmax chain measured in the
// eager path (profiles/, round 1).

#include "common.h"

#define CE_BLOCK 256

// ---------------- DINO CE forward ----------------
// x: [S, B, K] bf16 student logits ; t: [T, B, K] fp32 teacher probs.
// One block per student row (s, b). Outputs per row: lse, st (=sum of tsum),
// dot; loss_sum accumulated by atomicAdd: sum_rows (st*lse - dot).

__global__ void dino_ce_fwd_kernel(
    const __hip_bfloat16* __restrict__ x, const float* __restrict__ t,
    float* __restrict__ lse_out, float* __restrict__ st_out,
    float* __restrict__ loss_sum, int S, int T, int B, long K, float inv_temp,
    bool ignore_diag) {
  __shared__ float red[16];
  const int row = blockIdx.x;  // s * B + b
  const int s = row / B;
  const int b = row % B;
  const __hip_bfloat16* xr = x + (long)row * K;

  // pass 1: max of x/temp
  float m = -INFINITY;
  for (long i = threadIdx.x; i < K; i += blockDim.x) {
    m = fmaxf(m, bf16_to_f32(*(const short*)(xr + i)) * inv_temp);
  }
  m = block_reduce_max(m, red);
  __syncthreads();

  // pass 2: sum exp + teacher-sum dot
  float sumexp = 0.f, dot = 0.f, st = 0.f;
  for (long i = threadIdx.x; i < K; i += blockDim.x) {
    const float xi = bf16_to_f32(*(const short*)(xr + i)) * inv_temp;
    sumexp += __expf(xi - m);
    float ts = 0.f;
    for (int tt = 0; tt < T; ++tt) {
      if (ignore_diag && tt == s) continue;
      ts += t[((long)tt * B + b) * K + i];
    }
    dot += ts * xi;
    st += ts;
  }
  sumexp = block_reduce_sum(sumexp, red);
  __syncthreads();
  dot = block_reduce_sum(dot, red);
  __syncthreads();
  st = block_reduce_sum(st, red);
  const float lse = m + __logf(sumexp);
  if (threadIdx.x == 0) {
    lse_out[row] = lse;
    st_out[row] = st;
    atomicAdd(loss_sum, st * lse - dot);
  }
}

// backward: dx = g * inv_temp * (st * softmax - tsum)
__global__ void dino_ce_bwd_kernel(
    const __hip_bfloat16* __restrict__ x, const float* __restrict__ t,
    const float* __restrict__ lse_in, const float* __restrict__ st_in,
    const float* __restrict__ g, __hip_bfloat16* __restrict__ dx,
    int S, int T, int B, long K, float inv_temp, bool ignore_diag) {
  const int row = blockIdx.x;
  const int s = row / B;
  const int b = row % B;
  const __hip_bfloat16* xr = x + (long)row * K;
  __hip_bfloat16* dxr = dx + (long)row * K;
  const float lse = lse_in[row];
  const float st = st_in[row];
  const float scale = g[0] * inv_temp;
  for (long i = threadIdx.x; i < K; i += blockDim.x) {
    const float xi = bf16_to_f32(*(const short*)(xr + i)) * inv_temp;
    const float sm = __expf(xi - lse);
    float ts = 0.f;
    for (int tt = 0; tt < T; ++tt) {
      if (ignore_diag && tt == s) continue;
      ts += t[((long)tt * B + b) * K + i];
    }
    *reinterpret_cast<short*>(dxr + i) = f32_to_bf16(scale * (st * sm - ts));
  }
}

// ---------------- iBOT CE forward (row-aligned teacher) ----------------
// x: [M, K] bf16 ; t: [M, K] fp32 ; w: [M] fp32 per-row weight.
// loss_sum = sum_r w_r * (st_r * lse_r - dot_r)

__global__ void ibot_ce_fwd_kernel(
    const __hip_bfloat16* __restrict__ x, const float* __restrict__ t,
    const float* __restrict__ w, float* __restrict__ lse_out,
    float* __restrict__ st_out, float* __restrict__ loss_sum, long K,
    float inv_temp) {
  __shared__ float red[16];
  const int row = blockIdx.x;
  const __hip_bfloat16* xr = x + (long)row * K;
  const float* tr = t + (long)row * K;

  float m = -INFINITY;
  for (long i = threadIdx.x; i < K; i += blockDim.x) {
    m = fmaxf(m, bf16_to_f32(*(const short*)(xr + i)) * inv_temp);
  }
  m = block_reduce_max(m, red);
  __syncthreads();

  float sumexp = 0.f, dot = 0.f, st = 0.f;
  for (long i = threadIdx.x; i < K; i += blockDim.x) {
    const float xi = bf16_to_f32(*(const short*)(xr + i)) * inv_temp;
    sumexp += __expf(xi - m);
    const float ts = tr[i];
    dot += ts * xi;
    st += ts;
  }
  sumexp = block_reduce_sum(sumexp, red);
  __syncthreads();
  dot = block_reduce_sum(dot, red);
  __syncthreads();
  st = block_reduce_sum(st, red);
  const float lse = m + __logf(sumexp);
  if (threadIdx.x == 0) {
    lse_out[row] = lse;
    st_out[row] = st;
    atomicAdd(loss_sum, w[row] * (st * lse - dot));
  }
}

__global__ void ibot_ce_bwd_kernel(
    const __hip_bfloat16* __restrict__ x, const float* __restrict__ t,
    const float* __restrict__ w, const float* __restrict__ lse_in,
    const float* __restrict__ st_in, const float* __restrict__ g,
    __hip_bfloat16* __restrict__ dx, long K, float inv_temp) {
  const int row = blockIdx.x;
  const __hip_bfloat16* xr = x + (long)row * K;
  const float* tr = t + (long)row * K;
  __hip_bfloat16* dxr = dx + (long)row * K;
  const float lse = lse_in[row];
  const float st = st_in[row];
  const float scale = g[0] * inv_temp * w[row];
  for (long i = threadIdx.x; i < K; i += blockDim.x) {
    const float xi = bf16_to_f32(*(const short*)(xr + i)) * inv_temp;
    const float sm = __expf(xi - lse);
    *reinterpret_cast<short*>(dxr + i) = f32_to_bf16(scale * (st * sm - tr[i]));
  }
}

// ---------------- fused Sinkhorn-Knopp kernels ----------------
// Q is [M, K] fp32 (row-major; the reference works on Q^T [K, M] — we keep
// [M, K] and swap the roles: "rows" of the reference = our columns).
//
// exp kernel: Q = exp(x/temp) (bf16 in, fp32 out) + global sum via atomics.
__global__ void sinkhorn_exp_kernel(const __hip_bfloat16* __restrict__ x,
                                    float* __restrict__ Q, float* __restrict__ total,
                                    long n, float inv_temp) {
  __shared__ float red[16];
  float acc = 0.f;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x) {
    float v = __expf(bf16_to_f32(*(const short*)(x + i)) * inv_temp);
    Q[i] = v;
    acc += v;
  }
  acc = block_reduce_sum(acc, red);
  if (threadIdx.x == 0) atomicAdd(total, acc);
}

// column-sum over M rows (the reference's "sum of rows" of Q^T): out[k] =
// sum_m Q[m, k] / scale. One thread per column chunk, grid-stride.
__global__ void sinkhorn_colsum_kernel(const float* __restrict__ Q,
                                       float* __restrict__ out, int M, long K,
                                       const float* __restrict__ divisor) {
  const float inv = 1.0f / divisor[0];
  for (long k = blockIdx.x * (long)blockDim.x + threadIdx.x; k < K;
       k += (long)gridDim.x * blockDim.x) {
    float acc = 0.f;
    for (int m = 0; m < M; ++m) acc += Q[(long)m * K + k];
    out[k] = acc * inv;
  }
}

// divide columns by col vector (after RCCL all-reduce) and by K, then
// row-normalize: per row m: Q[m,k] /= (col[k] * K); rowsum -> Q[m,:] /= (rowsum * B)
// Final iteration multiplies by B instead (handled by flags).
__global__ void sinkhorn_div_row_kernel(float* __restrict__ Q,
                                        const float* __restrict__ col, int M, long K,
                                        float k_div, const float* __restrict__ B,
                                        bool scale_back) {
  __shared__ float red[16];
  const int m = blockIdx.x;
  float* qr = Q + (long)m * K;
  float acc = 0.f;
  for (long k = threadIdx.x; k < K; k += blockDim.x) {
    float v = qr[k] / (col[k] * k_div);
    qr[k] = v;
    acc += v;
  }
  acc = block_reduce_sum(acc, red);
  const float denom = acc * B[0];
  const float mul = scale_back ? (B[0] / denom) : (1.0f / denom);
  for (long k = threadIdx.x; k < K; k += blockDim.x) qr[k] *= mul;
}

// ---------------- factored Sinkhorn-Knopp ----------------
// Sinkhorn is diagonal scaling: Q_final = exp(x/T) ∘ u ⊗ v. Iterating on the
// [M] and [K] scale vectors only costs 2 read passes of the bf16 logits per
// iteration and never materializes the [M, 65536] fp32 Q (profiles/: the
// materialized version spent ~7 ms/step on Q traffic). Per-iteration scalar
// factors cancel in the final row-normalize, so no 1/sum_Q pass is needed.

// A[k] = sum_m exp(x[m,k]*inv_temp) * u[m]   (u == nullptr -> 1)
// 2D tiling: block (kw, mt) reduces an m_tile-row slab over a 2048-col window
// (contiguous 4 KB per row per block: streams, unlike a per-thread column
// walk with a 128 KB stride), then one atomicAdd per column per m-tile.
// m_tile is chosen by the launcher so kw*mt fills the 256 CUs even at the
// DINO shape (M=128, kw=32 -> m_tile 8 gives 512 blocks, not 32).
__global__ void sinkhorn_fact_colsum_kernel(const __hip_bfloat16* __restrict__ x,
                                            const float* __restrict__ u,
                                            float* __restrict__ A, int M, long K,
                                            float inv_temp, int m_tile) {
  const long k0 = (long)blockIdx.x * (blockDim.x * 8);
  const int m0 = blockIdx.y * m_tile;
  const int m1 = min(m0 + m_tile, M);
  float acc[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
  const long kb = k0 + threadIdx.x * 8;
  if (kb >= K) return;
  for (int m = m0; m < m1; ++m) {
    const float um = (u != nullptr ? u[m] : 1.0f);
    __hip_bfloat16 xb[8];
    Vec8<__hip_bfloat16>::load(xb, x + (long)m * K + kb);
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      acc[e] += um * __expf(bf16_to_f32(*(short*)(xb + e)) * inv_temp);
    }
  }
  if (gridDim.y == 1) {
#pragma unroll
    for (int e = 0; e < 8; ++e) A[kb + e] = acc[e];
  } else {
#pragma unroll
    for (int e = 0; e < 8; ++e) atomicAdd(A + kb + e, acc[e]);
  }
}

// u[m] = 1 / sum_k exp(x[m,k]*inv_temp) * v[k]
// k-split over gridDim.y so small-M launches (DINO: M=128) still fill the
// chip; partials land in sums[m] via atomicAdd and a finalize kernel inverts.
__global__ void sinkhorn_fact_rowsum_kernel(const __hip_bfloat16* __restrict__ x,
                                            const float* __restrict__ v,
                                            float* __restrict__ sums, long K,
                                            float inv_temp) {
  __shared__ float red[16];
  const int m = blockIdx.x;
  const int split = gridDim.y;
  const long kchunk = (K + split - 1) / split;
  const long ks = blockIdx.y * kchunk;
  const long ke = min(ks + kchunk, K);
  const __hip_bfloat16* xr = x + (long)m * K;
  float acc = 0.f;
  for (long k = ks + threadIdx.x; k < ke; k += blockDim.x) {
    acc += __expf(bf16_to_f32(*(const short*)(xr + k)) * inv_temp) * v[k];
  }
  acc = block_reduce_sum(acc, red);
  if (threadIdx.x == 0) {
    if (split == 1) sums[m] = 1.0f / acc;
    else atomicAdd(sums + m, acc);
  }
}

__global__ void sinkhorn_rowsum_invert_kernel(float* __restrict__ u, int M) {
  const int m = blockIdx.x * blockDim.x + threadIdx.x;
  if (m < M) u[m] = 1.0f / u[m];
}

// ---------------- CE with factored teacher ----------------
// t[m,k] = exp(xt[m,k]*inv_tt) * u[m] * v[k]; rows sum to 1 by construction.

__global__ void ibot_ce_fact_fwd_kernel(
    const __hip_bfloat16* __restrict__ x, const __hip_bfloat16* __restrict__ xt,
    const float* __restrict__ u, const float* __restrict__ v,
    const float* __restrict__ w, float* __restrict__ lse_out, float* __restrict__ st_out,
    float* __restrict__ loss_sum, long K, float inv_temp, float inv_tt) {
  __shared__ float red[16];
  const int row = blockIdx.x;
  const __hip_bfloat16* xr = x + (long)row * K;
  const __hip_bfloat16* xtr = xt + (long)row * K;
  const float um = u[row];

  float m = -INFINITY;
  for (long i = threadIdx.x; i < K; i += blockDim.x)
    m = fmaxf(m, bf16_to_f32(*(const short*)(xr + i)) * inv_temp);
  m = block_reduce_max(m, red);
  __syncthreads();

  float sumexp = 0.f, dot = 0.f, st = 0.f;
  for (long i = threadIdx.x; i < K; i += blockDim.x) {
    const float xi = bf16_to_f32(*(const short*)(xr + i)) * inv_temp;
    sumexp += __expf(xi - m);
    const float ts = __expf(bf16_to_f32(*(const short*)(xtr + i)) * inv_tt) * um * v[i];
    dot += ts * xi;
    st += ts;
  }
  sumexp = block_reduce_sum(sumexp, red);
  __syncthreads();
  dot = block_reduce_sum(dot, red);
  __syncthreads();
  st = block_reduce_sum(st, red);
  const float lse = m + __logf(sumexp);
  if (threadIdx.x == 0) {
    lse_out[row] = lse;
    st_out[row] = st;
    atomicAdd(loss_sum, w[row] * (st * lse - dot));
  }
}

__global__ void ibot_ce_fact_bwd_kernel(
    const __hip_bfloat16* __restrict__ x, const __hip_bfloat16* __restrict__ xt,
    const float* __restrict__ u, const float* __restrict__ v,
    const float* __restrict__ w, const float* __restrict__ lse_in,
    const float* __restrict__ st_in, const float* __restrict__ g,
    __hip_bfloat16* __restrict__ dx, long K, float inv_temp, float inv_tt) {
  const int row = blockIdx.x;
  const __hip_bfloat16* xr = x + (long)row * K;
  const __hip_bfloat16* xtr = xt + (long)row * K;
  const float um = u[row];
  const float lse = lse_in[row];
  const float st = st_in[row];
  const float scale = g[0] * inv_temp * w[row];
  __hip_bfloat16* dxr = dx + (long)row * K;
  for (long i = threadIdx.x; i < K; i += blockDim.x) {
    const float xi = bf16_to_f32(*(const short*)(xr + i)) * inv_temp;
    const float sm = __expf(xi - lse);
    const float ts = __expf(bf16_to_f32(*(const short*)(xtr + i)) * inv_tt) * um * v[i];
    *reinterpret_cast<short*>(dxr + i) = f32_to_bf16(scale * (st * sm - ts));
  }
}

__global__ void dino_ce_fact_fwd_kernel(
    const __hip_bfloat16* __restrict__ x, const __hip_bfloat16* __restrict__ xt,
    const float* __restrict__ u, const float* __restrict__ v,
    float* __restrict__ lse_out, float* __restrict__ st_out, float* __restrict__ loss_sum,
    int S, int T, int B, long K, float inv_temp, float inv_tt, bool ignore_diag) {
  __shared__ float red[16];
  const int row = blockIdx.x;
  const int s = row / B;
  const int b = row % B;
  const __hip_bfloat16* xr = x + (long)row * K;

  float m = -INFINITY;
  for (long i = threadIdx.x; i < K; i += blockDim.x)
    m = fmaxf(m, bf16_to_f32(*(const short*)(xr + i)) * inv_temp);
  m = block_reduce_max(m, red);
  __syncthreads();

  float sumexp = 0.f, dot = 0.f, st = 0.f;
  for (long i = threadIdx.x; i < K; i += blockDim.x) {
    const float xi = bf16_to_f32(*(const short*)(xr + i)) * inv_temp;
    sumexp += __expf(xi - m);
    float ts = 0.f;
    for (int tt = 0; tt < T; ++tt) {
      if (ignore_diag && tt == s) continue;
      const long trow = (long)tt * B + b;
      ts += __expf(bf16_to_f32(*(const short*)(xt + trow * K + i)) * inv_tt) * u[trow] * v[i];
    }
    dot += ts * xi;
    st += ts;
  }
  sumexp = block_reduce_sum(sumexp, red);
  __syncthreads();
  dot = block_reduce_sum(dot, red);
  __syncthreads();
  st = block_reduce_sum(st, red);
  const float lse = m + __logf(sumexp);
  if (threadIdx.x == 0) {
    lse_out[row] = lse;
    st_out[row] = st;
    atomicAdd(loss_sum, st * lse - dot);
  }
}

__global__ void dino_ce_fact_bwd_kernel(
    const __hip_bfloat16* __restrict__ x, const __hip_bfloat16* __restrict__ xt,
    const float* __restrict__ u, const float* __restrict__ v,
    const float* __restrict__ lse_in, const float* __restrict__ st_in,
    const float* __restrict__ g, __hip_bfloat16* __restrict__ dx, int S, int T, int B,
    long K, float inv_temp, float inv_tt, bool ignore_diag) {
  const int row = blockIdx.x;
  const int s = row / B;
  const int b = row % B;
  const __hip_bfloat16* xr = x + (long)row * K;
  __hip_bfloat16* dxr = dx + (long)row * K;
  const float lse = lse_in[row];
  const float st = st_in[row];
  const float scale = g[0] * inv_temp;
  for (long i = threadIdx.x; i < K; i += blockDim.x) {
    const float xi = bf16_to_f32(*(const short*)(xr + i)) * inv_temp;
    const float sm = __expf(xi - lse);
    float ts = 0.f;
    for (int tt = 0; tt < T; ++tt) {
      if (ignore_diag && tt == s) continue;
      const long trow = (long)tt * B + b;
      ts += __expf(bf16_to_f32(*(const short*)(xt + trow * K + i)) * inv_tt) * u[trow] * v[i];
    }
    *reinterpret_cast<short*>(dxr + i) = f32_to_bf16(scale * (st * sm - ts));
  }
}

// ---------------- launchers ----------------

void launch_dino_ce_fwd(const __hip_bfloat16* x, const float* t, float* lse, float* st,
                        float* loss_sum, int S, int T, int B, long K, float inv_temp,
                        bool ignore_diag, hipStream_t stream) {
  hipLaunchKernelGGL(dino_ce_fwd_kernel, dim3(S * B), dim3(CE_BLOCK), 0, stream, x, t,
                     lse, st, loss_sum, S, T, B, K, inv_temp, ignore_diag);
}

void launch_dino_ce_bwd(const __hip_bfloat16* x, const float* t, const float* lse,
                        const float* st, const float* g, __hip_bfloat16* dx, int S, int T,
                        int B, long K, float inv_temp, bool ignore_diag,
                        hipStream_t stream) {
  hipLaunchKernelGGL(dino_ce_bwd_kernel, dim3(S * B), dim3(CE_BLOCK), 0, stream, x, t,
                     lse, st, g, dx, S, T, B, K, inv_temp, ignore_diag);
}

void launch_ibot_ce_fwd(const __hip_bfloat16* x, const float* t, const float* w,
                        float* lse, float* st, float* loss_sum, int M, long K,
                        float inv_temp, hipStream_t stream) {
  if (M == 0) return;
  hipLaunchKernelGGL(ibot_ce_fwd_kernel, dim3(M), dim3(CE_BLOCK), 0, stream, x, t, w,
                     lse, st, loss_sum, K, inv_temp);
}

void launch_ibot_ce_bwd(const __hip_bfloat16* x, const float* t, const float* w,
                        const float* lse, const float* st, const float* g,
                        __hip_bfloat16* dx, int M, long K, float inv_temp,
                        hipStream_t stream) {
  if (M == 0) return;
  hipLaunchKernelGGL(ibot_ce_bwd_kernel, dim3(M), dim3(CE_BLOCK), 0, stream, x, t, w,
                     lse, st, g, dx, K, inv_temp);
}

void launch_sinkhorn_fact_colsum(const __hip_bfloat16* x, const float* u, float* A,
                                 int M, long K, float inv_temp, hipStream_t stream) {
  const int kw = (int)((K + CE_BLOCK * 8 - 1) / (CE_BLOCK * 8));
  // target >=512 workgroups so the 256-CU chip is filled even at M=128
  int tiles = 512 / (kw > 0 ? kw : 1);
  if (tiles < 1) tiles = 1;
  if (tiles > M) tiles = M;
  const int m_tile = (M + tiles - 1) / tiles;
  const int mt = (M + m_tile - 1) / m_tile;
  if (mt > 1) (void)hipMemsetAsync(A, 0, K * sizeof(float), stream);
  hipLaunchKernelGGL(sinkhorn_fact_colsum_kernel, dim3(kw, mt), dim3(CE_BLOCK), 0, stream,
                     x, u, A, M, K, inv_temp, m_tile);
}

void launch_sinkhorn_fact_rowsum(const __hip_bfloat16* x, const float* v, float* u,
                                 int M, long K, float inv_temp, hipStream_t stream) {
  int split = 512 / (M > 0 ? M : 1);
  if (split < 1) split = 1;
  if (split > 16) split = 16;
  if (split > 1) (void)hipMemsetAsync(u, 0, M * sizeof(float), stream);
  hipLaunchKernelGGL(sinkhorn_fact_rowsum_kernel, dim3(M, split), dim3(CE_BLOCK), 0,
                     stream, x, v, u, K, inv_temp);
  if (split > 1) {
    hipLaunchKernelGGL(sinkhorn_rowsum_invert_kernel, dim3((M + 255) / 256), dim3(256),
                       0, stream, u, M);
  }
}

void launch_ibot_ce_fact_fwd(const __hip_bfloat16* x, const __hip_bfloat16* xt,
                             const float* u, const float* v, const float* w, float* lse,
                             float* st, float* loss_sum, int M, long K, float inv_temp,
                             float inv_tt, hipStream_t stream) {
  if (M == 0) return;
  hipLaunchKernelGGL(ibot_ce_fact_fwd_kernel, dim3(M), dim3(CE_BLOCK), 0, stream, x, xt,
                     u, v, w, lse, st, loss_sum, K, inv_temp, inv_tt);
}

void launch_ibot_ce_fact_bwd(const __hip_bfloat16* x, const __hip_bfloat16* xt,
                             const float* u, const float* v, const float* w,
                             const float* lse, const float* st, const float* g,
                             __hip_bfloat16* dx, int M, long K, float inv_temp,
                             float inv_tt, hipStream_t stream) {
  if (M == 0) return;
  hipLaunchKernelGGL(ibot_ce_fact_bwd_kernel, dim3(M), dim3(CE_BLOCK), 0, stream, x, xt,
                     u, v, w, lse, st, g, dx, K, inv_temp, inv_tt);
}

void launch_dino_ce_fact_fwd(const __hip_bfloat16* x, const __hip_bfloat16* xt,
                             const float* u, const float* v, float* lse, float* st,
                             float* loss_sum, int S, int T, int B, long K, float inv_temp,
                             float inv_tt, bool ignore_diag, hipStream_t stream) {
  hipLaunchKernelGGL(dino_ce_fact_fwd_kernel, dim3(S * B), dim3(CE_BLOCK), 0, stream, x,
                     xt, u, v, lse, st, loss_sum, S, T, B, K, inv_temp, inv_tt,
                     ignore_diag);
}

void launch_dino_ce_fact_bwd(const __hip_bfloat16* x, const __hip_bfloat16* xt,
                             const float* u, const float* v, const float* lse,
                             const float* st, const float* g, __hip_bfloat16* dx, int S,
                             int T, int B, long K, float inv_temp, float inv_tt,
                             bool ignore_diag, hipStream_t stream) {
  hipLaunchKernelGGL(dino_ce_fact_bwd_kernel, dim3(S * B), dim3(CE_BLOCK), 0, stream, x,
                     xt, u, v, lse, st, g, dx, S, T, B, K, inv_temp, inv_tt, ignore_diag);
}

void launch_sinkhorn_exp(const __hip_bfloat16* x, float* Q, float* total, long n,
                         float inv_temp, hipStream_t stream) {
  int grid = (int)min((n + CE_BLOCK - 1) / CE_BLOCK, (long)4096);
  hipLaunchKernelGGL(sinkhorn_exp_kernel, dim3(grid), dim3(CE_BLOCK), 0, stream, x, Q,
                     total, n, inv_temp);
}

void launch_sinkhorn_colsum(const float* Q, float* out, int M, long K,
                            const float* divisor, hipStream_t stream) {
  int grid = (int)min((K + CE_BLOCK - 1) / CE_BLOCK, (long)2048);
  hipLaunchKernelGGL(sinkhorn_colsum_kernel, dim3(grid), dim3(CE_BLOCK), 0, stream, Q,
                     out, M, K, divisor);
}

void launch_sinkhorn_div_row(float* Q, const float* col, int M, long K, float k_div,
                             const float* B, bool scale_back, hipStream_t stream) {
  hipLaunchKernelGGL(sinkhorn_div_row_kernel, dim3(M), dim3(CE_BLOCK), 0, stream, Q, col,
                     M, K, k_div, B, scale_back);
}
