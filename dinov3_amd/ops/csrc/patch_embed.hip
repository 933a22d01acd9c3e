// Patch-embedding GEMM with the patchify gather fused into the A-operand
// load: out[n, d] = sum_k rows[n, k] * W[d, k] + bias[d], where
// rows[n, k] reads NCHW input directly — k = (c, dy, dx) so each 16-element
// k-slice is one contiguous 32 B pixel run; neighbouring rows (consecutive
// patches along W) are adjacent runs, so the staging loads coalesce.
//
// Structure: 128x128 output tile per 256-thread block (4 waves, each
// 32 rows x 128 cols = 2x4 accumulators of v_mfma_f32_32x32x16_bf16), LDS
// double-purpose staging of the A rows and the W slice per K-step of 32.

#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8_pe;
typedef __attribute__((ext_vector_type(16))) float f32x16_pe;

#define PE_MFMA(a, b, c) __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0)

namespace patch_embed {

DEV_INLINE bf16x8_pe load8(const __hip_bfloat16* p) {
  return *reinterpret_cast<const bf16x8_pe*>(p);
}

DEV_INLINE int c_row(int r, int h) { return (r & 3) + 8 * (r >> 2) + 4 * h; }

// x: [B, C, H, W] bf16 ; w: [D, K] bf16 (K = C*P*P) ; bias: [D] bf16
// out: [B*np, D] bf16 with np = (H/P)*(W/P). P is the template patch size.
template <int P, int C>
__global__ __launch_bounds__(256) void fwd_kernel(
    const __hip_bfloat16* __restrict__ x, const __hip_bfloat16* __restrict__ w,
    const __hip_bfloat16* __restrict__ bias, __hip_bfloat16* __restrict__ out,
    int Bimg, int H, int W, int D) {
  constexpr int K = C * P * P;        // 768 for P=16, C=3
  constexpr int BK = 32;              // K-step (two 16-wide MFMA slices)
  constexpr int A_STRIDE = BK + 8;    // bf16 elems; 16B-aligned, conflict-padded
  const int wp = W / P;
  const long rows = (long)Bimg * (H / P) * wp;

  const int row0 = blockIdx.x * 128;
  const int col0 = blockIdx.y * 128;
  const int wave = threadIdx.x / 64;
  const int lane = threadIdx.x & 63;
  const int hhalf = lane >> 5;
  const int l31 = lane & 31;

  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  __hip_bfloat16* a_lds = reinterpret_cast<__hip_bfloat16*>(smem_raw);            // [128][A_STRIDE]
  __hip_bfloat16* b_lds = a_lds + 128 * A_STRIDE;                                 // [128][A_STRIDE]

  // accumulators: wave covers rows [wave*32, wave*32+32) x cols [0,128)
  f32x16_pe acc[4];
#pragma unroll
  for (int t = 0; t < 4; ++t) acc[t] = {};

  // staging work split: 256 threads stage 128 rows x 32 k of A (each thread
  // one (row, 16k-run)) and the same shape of W.
  const int s_row = threadIdx.x / 2;          // 0..127
  const int s_half = (threadIdx.x & 1) * 16;  // which 16-k run of the 32-k step

  for (int k0 = 0; k0 < K; k0 += BK) {
    __syncthreads();
    {
      // A: rows[row0 + s_row][k0 + s_half .. +16) — one contiguous 32 B run
      const long n = (long)row0 + s_row;
      bf16x8_pe v0{}, v1{};
      if (n < rows) {
        const int k = k0 + s_half;
        const int c = k / (P * P);
        const int dy = (k / P) % P;
        const int dx = k % P;                 // 0 (runs are dx-aligned: P==16)
        const int img = (int)(n / ((long)(H / P) * wp));
        const int pidx = (int)(n % ((long)(H / P) * wp));
        const int ph = pidx / wp;
        const int pw = pidx % wp;
        const __hip_bfloat16* src =
            x + (((long)img * C + c) * H + (ph * P + dy)) * W + pw * P + dx;
        v0 = load8(src);
        v1 = load8(src + 8);
      }
      *reinterpret_cast<bf16x8_pe*>(&a_lds[s_row * A_STRIDE + s_half]) = v0;
      *reinterpret_cast<bf16x8_pe*>(&a_lds[s_row * A_STRIDE + s_half + 8]) = v1;
      // B: W[col0 + s_row][k0 + s_half .. +16)
      const int d = col0 + s_row;
      bf16x8_pe w0{}, w1{};
      if (d < D) {
        const __hip_bfloat16* srcw = w + (long)d * K + k0 + s_half;
        w0 = load8(srcw);
        w1 = load8(srcw + 8);
      }
      *reinterpret_cast<bf16x8_pe*>(&b_lds[s_row * A_STRIDE + s_half]) = w0;
      *reinterpret_cast<bf16x8_pe*>(&b_lds[s_row * A_STRIDE + s_half + 8]) = w1;
    }
    __syncthreads();

#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {  // two 16-k MFMA slices per K-step
      // A fragment: A[i = wave*32 + l31][k = ks*16 + h*8 + e]
      bf16x8_pe af = load8(&a_lds[(wave * 32 + l31) * A_STRIDE + ks * 16 + hhalf * 8]);
#pragma unroll
      for (int t = 0; t < 4; ++t) {
        // B fragment: B[k][j = t*32 + l31] = W[col0 + t*32 + l31][k]
        bf16x8_pe bf = load8(&b_lds[(t * 32 + l31) * A_STRIDE + ks * 16 + hhalf * 8]);
        acc[t] = PE_MFMA(af, bf, acc[t]);
      }
    }
  }

  // epilogue: lane l31 = output column within the t-th 32-col tile; rows run
  // over c_row(r, hhalf) + wave*32 (the probe-verified C layout)
  {
#pragma unroll
    for (int t = 0; t < 4; ++t) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const long rn = (long)row0 + wave * 32 + c_row(r, hhalf);
        const int d = col0 + t * 32 + l31;
        if (rn < rows && d < D) {
          float v = acc[t][r] + bf16_to_f32(*(const short*)(bias + d));
          *reinterpret_cast<short*>(out + rn * (long)D + d) = f32_to_bf16(v);
        }
      }
    }
  }
}

// Generic-P variant (ViT-g/14 and high-res adapt configs): the 16-element
// k-runs are no longer dx-aligned pixel runs, so the A staging gathers
// per-element with a k < K bound (K zero-padded up to the 32-wide step).
// Same 128x128 MFMA tile structure as the P=16 kernel.
template <int C>
__global__ __launch_bounds__(256) void fwd_kernel_anyP(
    const __hip_bfloat16* __restrict__ x, const __hip_bfloat16* __restrict__ w,
    const __hip_bfloat16* __restrict__ bias, __hip_bfloat16* __restrict__ out,
    int Bimg, int H, int W, int P, int D) {
  const int K = C * P * P;
  constexpr int BK = 32;
  constexpr int A_STRIDE = BK + 8;
  const int wp = W / P;
  const long rows = (long)Bimg * (H / P) * wp;

  const int row0 = blockIdx.x * 128;
  const int col0 = blockIdx.y * 128;
  const int wave = threadIdx.x / 64;
  const int lane = threadIdx.x & 63;
  const int hhalf = lane >> 5;
  const int l31 = lane & 31;

  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  __hip_bfloat16* a_lds = reinterpret_cast<__hip_bfloat16*>(smem_raw);
  __hip_bfloat16* b_lds = a_lds + 128 * A_STRIDE;

  f32x16_pe acc[4];
#pragma unroll
  for (int t = 0; t < 4; ++t) acc[t] = {};

  const int s_row = threadIdx.x / 2;
  const int s_half = (threadIdx.x & 1) * 16;

  // per-row patch coordinates hoisted out of the K loop
  const long n = (long)row0 + s_row;
  const bool n_ok = n < rows;
  int img = 0, ph = 0, pw = 0;
  if (n_ok) {
    img = (int)(n / ((long)(H / P) * wp));
    const int pidx = (int)(n % ((long)(H / P) * wp));
    ph = pidx / wp;
    pw = pidx % wp;
  }

  const int Kpad = (K + BK - 1) / BK * BK;
  for (int k0 = 0; k0 < Kpad; k0 += BK) {
    __syncthreads();
    {
      __hip_bfloat16 a_stage[16];
#pragma unroll
      for (int e = 0; e < 16; ++e) {
        const int k = k0 + s_half + e;
        __hip_bfloat16 v = __hip_bfloat16(0.f);
        if (n_ok && k < K) {
          const int c = k / (P * P);
          const int dy = (k / P) % P;
          const int dx = k % P;
          v = x[(((long)img * C + c) * H + (ph * P + dy)) * W + pw * P + dx];
        }
        a_stage[e] = v;
      }
      *reinterpret_cast<bf16x8_pe*>(&a_lds[s_row * A_STRIDE + s_half]) =
          *reinterpret_cast<bf16x8_pe*>(&a_stage[0]);
      *reinterpret_cast<bf16x8_pe*>(&a_lds[s_row * A_STRIDE + s_half + 8]) =
          *reinterpret_cast<bf16x8_pe*>(&a_stage[8]);

      const int d = col0 + s_row;
      __hip_bfloat16 w_stage[16];
#pragma unroll
      for (int e = 0; e < 16; ++e) {
        const int k = k0 + s_half + e;
        w_stage[e] = (d < D && k < K) ? w[(long)d * K + k] : __hip_bfloat16(0.f);
      }
      *reinterpret_cast<bf16x8_pe*>(&b_lds[s_row * A_STRIDE + s_half]) =
          *reinterpret_cast<bf16x8_pe*>(&w_stage[0]);
      *reinterpret_cast<bf16x8_pe*>(&b_lds[s_row * A_STRIDE + s_half + 8]) =
          *reinterpret_cast<bf16x8_pe*>(&w_stage[8]);
    }
    __syncthreads();

#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      bf16x8_pe af = load8(&a_lds[(wave * 32 + l31) * A_STRIDE + ks * 16 + hhalf * 8]);
#pragma unroll
      for (int t = 0; t < 4; ++t) {
        bf16x8_pe bf = load8(&b_lds[(t * 32 + l31) * A_STRIDE + ks * 16 + hhalf * 8]);
        acc[t] = PE_MFMA(af, bf, acc[t]);
      }
    }
  }

  {
#pragma unroll
    for (int t = 0; t < 4; ++t) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const long rn = (long)row0 + wave * 32 + c_row(r, hhalf);
        const int d = col0 + t * 32 + l31;
        if (rn < rows && d < D) {
          float v = acc[t][r] + bf16_to_f32(*(const short*)(bias + d));
          *reinterpret_cast<short*>(out + rn * (long)D + d) = f32_to_bf16(v);
        }
      }
    }
  }
}

}  // namespace patch_embed

void launch_patch_embed_fwd(const __hip_bfloat16* x, const __hip_bfloat16* w,
                            const __hip_bfloat16* bias, __hip_bfloat16* out, int Bimg,
                            int C, int H, int W, int P, int D, hipStream_t stream) {
  const long rows = (long)Bimg * (H / P) * (W / P);
  dim3 grid((rows + 127) / 128, (D + 127) / 128);
  size_t shmem = 2 * 128 * (32 + 8) * sizeof(__hip_bfloat16);
  if (P == 16 && C == 3) {
    // fast path: 16-wide k-slices are dx-aligned contiguous pixel runs
    hipLaunchKernelGGL(HIP_KERNEL_NAME(patch_embed::fwd_kernel<16, 3>), grid, dim3(256),
                       shmem, stream, x, w, bias, out, Bimg, H, W, D);
  } else if (C == 3) {
    // generic patch size (14 for ViT-g/14, 8, ...): gather staging
    hipLaunchKernelGGL(HIP_KERNEL_NAME(patch_embed::fwd_kernel_anyP<3>), grid, dim3(256),
                       shmem, stream, x, w, bias, out, Bimg, H, W, P, D);
  }
}
