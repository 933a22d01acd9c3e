// Fused RoPE + multi-head attention on the raw QKV projection (K5 + K6).
//
// Reads qkv [B, N, 3, H, hd] bf16 (the qkv GEMM output, no permutes), applies
// rotate-half RoPE (fp32 sin/cos [P, hd] tables, prefix tokens pass through)
// to Q in-register and to K at LDS staging, writes O token-major [B, N, H, hd]
// so the out-projection GEMM consumes it directly. Backward writes dq/dk/dv
// into ONE dqkv buffer of the same layout and applies the inverse rotation to
// dQ/dK in the accumulator epilogues.
//
// Same MFMA machinery as fmha.hip (swapped S^T, permlane32_swap repack);
// see that file for the fragment-layout contract (probe-verified).

#include "common.h"
#include <cstdlib>

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(16))) float f32x16;

#define MFMA32(a, b, c) __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0)

namespace fmha_rope {

DEV_INLINE unsigned cvt_pk_bf16(float lo, float hi) {
  unsigned r;
  asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(lo), "v"(hi));
  return r;
}

DEV_INLINE void permlane32_swap(unsigned& a, unsigned& b) {
  auto r = __builtin_amdgcn_permlane32_swap(a, b, false, false);
  a = r[0];
  b = r[1];
}

DEV_INLINE bf16x8 pack_fragment(const float* s, int base) {
  unsigned r01 = cvt_pk_bf16(s[base + 0], s[base + 1]);
  unsigned r23 = cvt_pk_bf16(s[base + 2], s[base + 3]);
  unsigned r45 = cvt_pk_bf16(s[base + 4], s[base + 5]);
  unsigned r67 = cvt_pk_bf16(s[base + 6], s[base + 7]);
  permlane32_swap(r01, r45);
  permlane32_swap(r23, r67);
  union {
    unsigned u[4];
    bf16x8 v;
  } out;
  out.u[0] = r01;
  out.u[1] = r23;
  out.u[2] = r45;
  out.u[3] = r67;
  return out.v;
}

DEV_INLINE bf16x8 load8(const __hip_bfloat16* p) {
  return *reinterpret_cast<const bf16x8*>(p);
}

DEV_INLINE int c_row(int r, int h) { return (r & 3) + 8 * (r >> 2) + 4 * h; }

DEV_INLINE float b2f(__bf16 x) {
  union { float f; unsigned i; } v;
  v.i = ((unsigned)*(unsigned short*)&x) << 16;
  return v.f;
}

// rotate a (lo, hi) register pair of 8 bf16 each with fp32 tables at
// sin/cos row + column offset c0 (c0 < hd/2)
DEV_INLINE void rope_rotate8(bf16x8& lo, bf16x8& hi, const float* sinrow,
                             const float* cosrow, int c0) {
  union { unsigned u[4]; bf16x8 v; } lo_out, hi_out;
  float flo[8], fhi[8];
#pragma unroll
  for (int e = 0; e < 8; ++e) {
    const float c = cosrow[c0 + e];
    const float s = sinrow[c0 + e];
    const float l = b2f(lo[e]);
    const float h = b2f(hi[e]);
    flo[e] = l * c - h * s;
    fhi[e] = h * c + l * s;
  }
#pragma unroll
  for (int e = 0; e < 4; ++e) {
    lo_out.u[e] = cvt_pk_bf16(flo[2 * e], flo[2 * e + 1]);
    hi_out.u[e] = cvt_pk_bf16(fhi[2 * e], fhi[2 * e + 1]);
  }
  lo = lo_out.v;
  hi = hi_out.v;
}

// qkv addressing: element (n, which, d) for fixed (b, h)
// offset = ((((long)b*N + n)*3 + which)*H + h)*hd + d
struct QkvView {
  const __hip_bfloat16* base;  // qkv + (b, h) folded: base = qkv + (b*N*3*H + h)*hd
  long row_stride;             // 3*H*hd
  long which_stride;           // H*hd
  DEV_INLINE const __hip_bfloat16* at(int n, int which, int d) const {
    return base + (long)n * row_stride + which * which_stride + d;
  }
};

// ---------------------------------------------------------------------------
// Forward
// ---------------------------------------------------------------------------
// QLDS/PREFETCH: the hd-128 long-N configuration holds 304 registers/thread
// with in-register Q fragments + double-buffered K/V staging -> occupancy
// 1 wave/SIMD and fully exposed HBM latency (252 GB/s effective at N=2305).
// QLDS parks the rotated Q fragments in LDS in their REGISTER layout (lane-
// indexed, conflict-free b128), PREFETCH=false drops the register double
// buffer; together they fit 2 waves/SIMD.
template <int HD, int NT = 256, bool QLDS = false, bool PREFETCH = true, int MINW = 1,
          int KVB = 32>
__global__ __launch_bounds__(NT, MINW) void fwd_kernel(
    const __hip_bfloat16* __restrict__ qkv, const float* __restrict__ sin_t,
    const float* __restrict__ cos_t, __hip_bfloat16* __restrict__ o,
    float* __restrict__ lse, int B, int H, int N, int P, float scale) {
  constexpr int KSLICES = HD / 16;
  constexpr int DTILES = HD / 32;
  constexpr int LDS_STRIDE = HD + 8;
  constexpr int VT_STRIDE = 2 * KVB + 8;
  constexpr int HALF = HD / 2;

  const int bh = blockIdx.x;
  const int b = bh / H;
  const int h = bh % H;
  const int wave = threadIdx.x / 64;
  const int lane = threadIdx.x & 63;
  const int hhalf = lane >> 5;
  const int l31 = lane & 31;
  const int q0 = blockIdx.y * (NT / 2) + wave * 32;
  const int prefix = N - P;
  const bool use_rope = (sin_t != nullptr);

  QkvView qv;
  qv.base = qkv + ((long)b * N * 3 * H + h) * HD;
  qv.row_stride = (long)3 * H * HD;
  qv.which_stride = (long)H * HD;

  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  __hip_bfloat16* k_lds = reinterpret_cast<__hip_bfloat16*>(smem_raw);
  __hip_bfloat16* vt_lds = k_lds + 2 * KVB * LDS_STRIDE;
  // per-wave Q fragment store in register layout: index = slice, lane
  __hip_bfloat16* q_lds = vt_lds + HD * VT_STRIDE + (QLDS ? wave * KSLICES * 64 * 8 : 0);

  // Q fragments (+ rope)
  bf16x8 qf[QLDS ? 1 : KSLICES];
  {
    bf16x8 qtmp[KSLICES];
    const int qrow = q0 + l31;
    const int safe = qrow < N ? qrow : (N - 1);
#pragma unroll
    for (int s = 0; s < KSLICES; ++s) qtmp[s] = load8(qv.at(safe, 0, s * 16 + hhalf * 8));
    const int p = safe - prefix;
    if (use_rope && p >= 0) {
      const float* srow = sin_t + (long)p * HD;
      const float* crow = cos_t + (long)p * HD;
#pragma unroll
      for (int s2 = 0; s2 < KSLICES / 2; ++s2) {
        rope_rotate8(qtmp[s2], qtmp[s2 + KSLICES / 2], srow, crow, s2 * 16 + hhalf * 8);
      }
    }
    if constexpr (QLDS) {
#pragma unroll
      for (int s = 0; s < KSLICES; ++s)
        *reinterpret_cast<bf16x8*>(&q_lds[(s * 64 + lane) * 8]) = qtmp[s];
    } else {
#pragma unroll
      for (int s = 0; s < KSLICES; ++s) qf[s] = qtmp[s];
    }
  }

  float o_acc[DTILES][16];
#pragma unroll
  for (int t = 0; t < DTILES; ++t)
#pragma unroll
    for (int r = 0; r < 16; ++r) o_acc[t][r] = 0.f;
  float m_run = -INFINITY, l_run = 0.f;

  // T14 async-STAGE split: each thread owns ONE K pair-chunk and up to two V
  // chunks of the staged 64-key tile; the next tile's global loads are issued
  // BEFORE the compute phase so HBM latency hides under the MFMAs instead of
  // draining at the pre-compute barrier.
  constexpr int PAIRS_PER_ROW = HALF / 8;     // K pair-items per row
  constexpr int PER_ROW = HD / 8;             // V items per row
  constexpr int K_ITEMS = 2 * KVB * PAIRS_PER_ROW;   // 256 (hd64) / 512 (hd128)
  constexpr int K_ITEMS_PER_THREAD = (K_ITEMS + NT - 1) / NT;
  constexpr int V_ITEMS_PER_THREAD = (2 * KVB * PER_ROW + NT - 1) / NT;

  bf16x8 klo[K_ITEMS_PER_THREAD], khi[K_ITEMS_PER_THREAD];
  bf16x8 vreg[V_ITEMS_PER_THREAD];
  auto issue_loads = [&](int kbase0) {
#pragma unroll
    for (int j = 0; j < K_ITEMS_PER_THREAD; ++j) {
      const int item = threadIdx.x + j * NT;
      const int krow = kbase0 + item / PAIRS_PER_ROW;
      const int c0 = (item % PAIRS_PER_ROW) * 8;
      const bool ok = item < K_ITEMS && krow < N;
      klo[j] = ok ? load8(qv.at(krow, 1, c0)) : bf16x8{};
      khi[j] = ok ? load8(qv.at(krow, 1, c0 + HALF)) : bf16x8{};
    }
#pragma unroll
    for (int j = 0; j < V_ITEMS_PER_THREAD; ++j) {
      const int idx = threadIdx.x + j * NT;
      const int row = idx / PER_ROW;
      const int krow = kbase0 + row;
      vreg[j] = (idx < 2 * KVB * PER_ROW && krow < N) ? load8(qv.at(krow, 2, (idx % PER_ROW) * 8))
                                                      : bf16x8{};
    }
  };
  auto write_tile = [&](int kbase0) {
#pragma unroll
    for (int j = 0; j < K_ITEMS_PER_THREAD; ++j) {
      const int item = threadIdx.x + j * NT;
      if (item < K_ITEMS) {
        const int lrow = item / PAIRS_PER_ROW;
        const int c0 = (item % PAIRS_PER_ROW) * 8;
        const int krow = kbase0 + lrow;
        const int p = krow - prefix;
        if (use_rope && krow < N && p >= 0)
          rope_rotate8(klo[j], khi[j], sin_t + (long)p * HD, cos_t + (long)p * HD, c0);
        *reinterpret_cast<bf16x8*>(&k_lds[lrow * LDS_STRIDE + c0]) = klo[j];
        *reinterpret_cast<bf16x8*>(&k_lds[lrow * LDS_STRIDE + HALF + c0]) = khi[j];
      }
    }
#pragma unroll
    for (int j = 0; j < V_ITEMS_PER_THREAD; ++j) {
      const int idx = threadIdx.x + j * NT;
      if (idx < 2 * KVB * PER_ROW) {
        const int row = idx / PER_ROW;
        const int c8 = (idx % PER_ROW) * 8;
#pragma unroll
        for (int e = 0; e < 8; ++e)
          vt_lds[(c8 + e) * VT_STRIDE + row] = reinterpret_cast<__hip_bfloat16*>(&vreg[j])[e];
      }
    }
  };

  const int n_super = (N + 2 * KVB - 1) / (2 * KVB);  // 64 keys per barrier pair
  if constexpr (PREFETCH) issue_loads(0);
  for (int kt = 0; kt < n_super; ++kt) {
    const int kbase0 = kt * 2 * KVB;
    __syncthreads();               // compute of tile kt-1 done reading LDS
    if constexpr (!PREFETCH) issue_loads(kbase0);
    write_tile(kbase0);            // regs -> LDS (rope applied at write)
    if constexpr (PREFETCH) {
      if (kt + 1 < n_super) issue_loads(kbase0 + 2 * KVB);  // fly during compute
    }
    __syncthreads();

   constexpr int SUBK = 32;  // keys per MFMA tile (fixed by the 32x32 MFMA)
   for (int sub = 0; sub < (2 * KVB) / SUBK && kbase0 + sub * SUBK < N; ++sub) {
    const int kbase = kbase0 + sub * SUBK;
    const int krow_off = sub * SUBK;

    f32x16 s_acc = {};
#pragma unroll
    for (int s = 0; s < KSLICES; ++s) {
      bf16x8 af = load8(&k_lds[(krow_off + l31) * LDS_STRIDE + s * 16 + hhalf * 8]);
      bf16x8 qs;
      if constexpr (QLDS) qs = load8(&q_lds[(s * 64 + lane) * 8]);
      else qs = qf[s];
      s_acc = MFMA32(af, qs, s_acc);
    }
    float sv[16];
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int krow = kbase + c_row(r, hhalf);
      sv[r] = (krow < N) ? s_acc[r] * scale : -INFINITY;
    }
    float tmax = sv[0];
#pragma unroll
    for (int r = 1; r < 16; ++r) tmax = fmaxf(tmax, sv[r]);
    tmax = fmaxf(tmax, __shfl_xor(tmax, 32, 64));
    const float m_new = fmaxf(m_run, tmax);
    const float alpha = (m_run == -INFINITY) ? 0.f : __expf(m_run - m_new);
    float psum = 0.f;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      sv[r] = __expf(sv[r] - m_new);
      psum += sv[r];
    }
    psum += __shfl_xor(psum, 32, 64);
    l_run = l_run * alpha + psum;
    m_run = m_new;
#pragma unroll
    for (int t = 0; t < DTILES; ++t)
#pragma unroll
      for (int r = 0; r < 16; ++r) o_acc[t][r] *= alpha;

    bf16x8 p0 = pack_fragment(sv, 0);
    bf16x8 p1 = pack_fragment(sv, 8);
#pragma unroll
    for (int t = 0; t < DTILES; ++t) {
      bf16x8 a0 = load8(&vt_lds[(t * 32 + l31) * VT_STRIDE + krow_off + hhalf * 8]);
      bf16x8 a1 = load8(&vt_lds[(t * 32 + l31) * VT_STRIDE + krow_off + 16 + hhalf * 8]);
      f32x16 acc;
#pragma unroll
      for (int r = 0; r < 16; ++r) acc[r] = o_acc[t][r];
      acc = MFMA32(a0, p0, acc);
      acc = MFMA32(a1, p1, acc);
#pragma unroll
      for (int r = 0; r < 16; ++r) o_acc[t][r] = acc[r];
    }
   }
  }

  const float inv_l = 1.0f / l_run;
  const int qrow = q0 + l31;
  if (qrow < N) {
    // O token-major: ((b*N + n)*H + h)*hd + d
    __hip_bfloat16* op = o + (((long)b * N + qrow) * H + h) * HD;
#pragma unroll
    for (int t = 0; t < DTILES; ++t)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int d = t * 32 + c_row(r, hhalf);
        *reinterpret_cast<short*>(op + d) = f32_to_bf16(o_acc[t][r] * inv_l);
      }
    if (hhalf == 0) lse[((long)b * H + h) * N + qrow] = m_run + __logf(l_run);
  }
}

// ---------------------------------------------------------------------------
// Backward dQ
// ---------------------------------------------------------------------------
template <int HD, int NT = 256, int MINW = 1, int KVB = 32>
__global__ __launch_bounds__(NT, MINW) void bwd_dq_kernel(
    const __hip_bfloat16* __restrict__ qkv, const __hip_bfloat16* __restrict__ dout,
    const float* __restrict__ sin_t, const float* __restrict__ cos_t,
    const float* __restrict__ lse, const float* __restrict__ D,
    __hip_bfloat16* __restrict__ dqkv, int B, int H, int N, int P, float scale) {
  constexpr int KSLICES = HD / 16;
  constexpr int DTILES = HD / 32;
  constexpr int LDS_STRIDE = HD + 8;
  constexpr int KT_STRIDE = 2 * KVB + 8;
  constexpr int HALF = HD / 2;

  const int bh = blockIdx.x;
  const int b = bh / H;
  const int h = bh % H;
  const int wave = threadIdx.x / 64;
  const int lane = threadIdx.x & 63;
  const int hhalf = lane >> 5;
  const int l31 = lane & 31;
  const int q0 = blockIdx.y * (NT / 2) + wave * 32;
  const int prefix = N - P;
  const bool use_rope = (sin_t != nullptr);

  QkvView qv;
  qv.base = qkv + ((long)b * N * 3 * H + h) * HD;
  qv.row_stride = (long)3 * H * HD;
  qv.which_stride = (long)H * HD;
  const __hip_bfloat16* do_base = dout + ((long)b * N * H + h) * HD;  // token-major dO
  const long do_stride = (long)H * HD;

  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  __hip_bfloat16* k_lds = reinterpret_cast<__hip_bfloat16*>(smem_raw);   // rope'd K
  __hip_bfloat16* v_lds = k_lds + 2 * KVB * LDS_STRIDE;
  __hip_bfloat16* kt_lds = v_lds + 2 * KVB * LDS_STRIDE;                  // rope'd K^T

  const int qrow = q0 + l31;
  const int safe = qrow < N ? qrow : (N - 1);
  bf16x8 qf[KSLICES], dof[KSLICES];
  {
#pragma unroll
    for (int s = 0; s < KSLICES; ++s) {
      qf[s] = load8(qv.at(safe, 0, s * 16 + hhalf * 8));
      dof[s] = load8(do_base + (long)safe * do_stride + s * 16 + hhalf * 8);
    }
    const int p = safe - prefix;
    if (use_rope && p >= 0) {
      const float* srow = sin_t + (long)p * HD;
      const float* crow = cos_t + (long)p * HD;
#pragma unroll
      for (int s2 = 0; s2 < KSLICES / 2; ++s2)
        rope_rotate8(qf[s2], qf[s2 + KSLICES / 2], srow, crow, s2 * 16 + hhalf * 8);
    }
  }
  const float my_lse = lse[((long)b * H + h) * N + safe];
  const float my_D = D[((long)b * H + h) * N + safe];

  float dq_acc[DTILES][16];
#pragma unroll
  for (int t = 0; t < DTILES; ++t)
#pragma unroll
    for (int r = 0; r < 16; ++r) dq_acc[t][r] = 0.f;

  constexpr int PAIRS_PER_ROW = HALF / 8;
  constexpr int PER_ROW = HD / 8;
  constexpr int K_ITEMS = 2 * KVB * PAIRS_PER_ROW;
  constexpr int K_IPT = (K_ITEMS + NT - 1) / NT;
  constexpr int V_IPT = (2 * KVB * PER_ROW + NT - 1) / NT;
  bf16x8 klo[K_IPT], khi[K_IPT], vreg[V_IPT];
  auto issue_loads = [&](int kbase0) {
#pragma unroll
    for (int j = 0; j < K_IPT; ++j) {
      const int item = threadIdx.x + j * NT;
      const int krow = kbase0 + item / PAIRS_PER_ROW;
      const int c0 = (item % PAIRS_PER_ROW) * 8;
      const bool ok = item < K_ITEMS && krow < N;
      klo[j] = ok ? load8(qv.at(krow, 1, c0)) : bf16x8{};
      khi[j] = ok ? load8(qv.at(krow, 1, c0 + HALF)) : bf16x8{};
    }
#pragma unroll
    for (int j = 0; j < V_IPT; ++j) {
      const int idx = threadIdx.x + j * NT;
      const int krow = kbase0 + idx / PER_ROW;
      vreg[j] = (idx < 2 * KVB * PER_ROW && krow < N)
                    ? load8(qv.at(krow, 2, (idx % PER_ROW) * 8)) : bf16x8{};
    }
  };
  auto write_tile = [&](int kbase0) {
#pragma unroll
    for (int j = 0; j < K_IPT; ++j) {
      const int item = threadIdx.x + j * NT;
      if (item < K_ITEMS) {
        const int lrow = item / PAIRS_PER_ROW;
        const int c0 = (item % PAIRS_PER_ROW) * 8;
        const int krow = kbase0 + lrow;
        const int p = krow - prefix;
        if (use_rope && krow < N && p >= 0)
          rope_rotate8(klo[j], khi[j], sin_t + (long)p * HD, cos_t + (long)p * HD, c0);
        *reinterpret_cast<bf16x8*>(&k_lds[lrow * LDS_STRIDE + c0]) = klo[j];
        *reinterpret_cast<bf16x8*>(&k_lds[lrow * LDS_STRIDE + HALF + c0]) = khi[j];
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          kt_lds[(c0 + e) * KT_STRIDE + lrow] = reinterpret_cast<__hip_bfloat16*>(&klo[j])[e];
          kt_lds[(c0 + HALF + e) * KT_STRIDE + lrow] = reinterpret_cast<__hip_bfloat16*>(&khi[j])[e];
        }
      }
    }
#pragma unroll
    for (int j = 0; j < V_IPT; ++j) {
      const int idx = threadIdx.x + j * NT;
      if (idx < 2 * KVB * PER_ROW)
        *reinterpret_cast<bf16x8*>(&v_lds[(idx / PER_ROW) * LDS_STRIDE + (idx % PER_ROW) * 8]) = vreg[j];
    }
  };

  const int n_super = (N + 2 * KVB - 1) / (2 * KVB);
  issue_loads(0);
  for (int kt = 0; kt < n_super; ++kt) {
    const int kbase0 = kt * 2 * KVB;
    __syncthreads();
    write_tile(kbase0);
    if (kt + 1 < n_super) issue_loads(kbase0 + 2 * KVB);
    __syncthreads();

   constexpr int SUBK = 32;  // keys per MFMA tile (fixed by the 32x32 MFMA)
   for (int sub = 0; sub < (2 * KVB) / SUBK && kbase0 + sub * SUBK < N; ++sub) {
    const int kbase = kbase0 + sub * SUBK;
    const int krow_off = sub * SUBK;
    f32x16 s_acc = {}, dp_acc = {};
#pragma unroll
    for (int s = 0; s < KSLICES; ++s) {
      bf16x8 kf = load8(&k_lds[(krow_off + l31) * LDS_STRIDE + s * 16 + hhalf * 8]);
      bf16x8 vf = load8(&v_lds[(krow_off + l31) * LDS_STRIDE + s * 16 + hhalf * 8]);
      s_acc = MFMA32(kf, qf[s], s_acc);
      dp_acc = MFMA32(vf, dof[s], dp_acc);
    }
    float ds[16];
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int krow = kbase + c_row(r, hhalf);
      float p = (krow < N && qrow < N) ? __expf(s_acc[r] * scale - my_lse) : 0.f;
      ds[r] = p * (dp_acc[r] - my_D) * scale;
    }
    bf16x8 f0 = pack_fragment(ds, 0);
    bf16x8 f1 = pack_fragment(ds, 8);
#pragma unroll
    for (int t = 0; t < DTILES; ++t) {
      bf16x8 a0 = load8(&kt_lds[(t * 32 + l31) * KT_STRIDE + krow_off + hhalf * 8]);
      bf16x8 a1 = load8(&kt_lds[(t * 32 + l31) * KT_STRIDE + krow_off + 16 + hhalf * 8]);
      f32x16 acc;
#pragma unroll
      for (int r = 0; r < 16; ++r) acc[r] = dq_acc[t][r];
      acc = MFMA32(a0, f0, acc);
      acc = MFMA32(a1, f1, acc);
#pragma unroll
      for (int r = 0; r < 16; ++r) dq_acc[t][r] = acc[r];
    }
   }
  }

  if (qrow < N) {
    // inverse rope on dQ: pairs (d, d+HALF) = (tile t, tile t+DTILES/2) reg r
    const int p = qrow - prefix;
    if (use_rope && p >= 0) {
      const float* srow = sin_t + (long)p * HD;
      const float* crow = cos_t + (long)p * HD;
#pragma unroll
      for (int t = 0; t < DTILES / 2; ++t)
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int d = t * 32 + c_row(r, hhalf);
          const float c = crow[d], s = srow[d];
          const float glo = dq_acc[t][r], ghi = dq_acc[t + DTILES / 2][r];
          dq_acc[t][r] = glo * c + ghi * s;
          dq_acc[t + DTILES / 2][r] = ghi * c - glo * s;
        }
    }
    __hip_bfloat16* dqp = dqkv + ((((long)b * N + qrow) * 3 + 0) * H + h) * HD;
#pragma unroll
    for (int t = 0; t < DTILES; ++t)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int d = t * 32 + c_row(r, hhalf);
        *reinterpret_cast<short*>(dqp + d) = f32_to_bf16(dq_acc[t][r]);
      }
  }
}

// ---------------------------------------------------------------------------
// Backward dK/dV
// ---------------------------------------------------------------------------
// MODE splits the kernel for register-starved configs (hd-128 holds 366
// regs with both accumulator sets -> occupancy 1): MODE=1 computes only dV
// (no vf/dP/dK state), MODE=2 only dK. Each re-streams Q/dO, but at 2
// waves/SIMD instead of 1 the latency hiding more than pays for it.
template <int HD, int NT = 256, int MINW = 1, int MODE = 0>
__global__ __launch_bounds__(NT, MINW) void bwd_dkv_kernel(
    const __hip_bfloat16* __restrict__ qkv, const __hip_bfloat16* __restrict__ dout,
    const float* __restrict__ sin_t, const float* __restrict__ cos_t,
    const float* __restrict__ lse, const float* __restrict__ D,
    __hip_bfloat16* __restrict__ dqkv, int B, int H, int N, int P, float scale) {
  constexpr int KSLICES = HD / 16;
  constexpr int DTILES = HD / 32;
  constexpr int QB = 32;
  constexpr int QT_STRIDE = QB + 8;
  constexpr int HALF = HD / 2;

  const int bh = blockIdx.x;
  const int b = bh / H;
  const int h = bh % H;
  const int wave = threadIdx.x / 64;
  const int lane = threadIdx.x & 63;
  const int hhalf = lane >> 5;
  const int l31 = lane & 31;
  const int k0 = blockIdx.y * 128 + wave * 32;
  const int prefix = N - P;
  const bool use_rope = (sin_t != nullptr);

  QkvView qv;
  qv.base = qkv + ((long)b * N * 3 * H + h) * HD;
  qv.row_stride = (long)3 * H * HD;
  qv.which_stride = (long)H * HD;
  const __hip_bfloat16* do_base = dout + ((long)b * N * H + h) * HD;
  const long do_stride = (long)H * HD;

  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  __hip_bfloat16* qt_lds = reinterpret_cast<__hip_bfloat16*>(smem_raw);  // rope'd Q^T
  __hip_bfloat16* dot_lds = qt_lds + HD * QT_STRIDE;                      // dO^T
  float* lse_lds = reinterpret_cast<float*>(dot_lds + HD * QT_STRIDE);
  float* d_lds = lse_lds + QB;

  const int krow = k0 + l31;
  const int safe = krow < N ? krow : (N - 1);
  bf16x8 kf[KSLICES], vf[MODE == 1 ? 1 : KSLICES];
  {
#pragma unroll
    for (int s = 0; s < KSLICES; ++s) {
      kf[s] = load8(qv.at(safe, 1, s * 16 + hhalf * 8));
      if constexpr (MODE != 1) vf[s] = load8(qv.at(safe, 2, s * 16 + hhalf * 8));
    }
    const int p = safe - prefix;
    if (use_rope && p >= 0) {
      const float* srow = sin_t + (long)p * HD;
      const float* crow = cos_t + (long)p * HD;
#pragma unroll
      for (int s2 = 0; s2 < KSLICES / 2; ++s2)
        rope_rotate8(kf[s2], kf[s2 + KSLICES / 2], srow, crow, s2 * 16 + hhalf * 8);
    }
  }

  float dk_acc[MODE == 1 ? 1 : DTILES][16], dv_acc[MODE == 2 ? 1 : DTILES][16];
#pragma unroll
  for (int t = 0; t < DTILES; ++t)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      if constexpr (MODE != 1) dk_acc[t][r] = 0.f;
      if constexpr (MODE != 2) dv_acc[t][r] = 0.f;
    }

  constexpr int PAIRS_PER_ROW = HALF / 8;
  constexpr int PER_ROW = HD / 8;
  constexpr int QK_ITEMS = QB * PAIRS_PER_ROW;   // <= 256
  constexpr int DO_ITEMS = QB * PER_ROW;         // <= 512
  // QK_ITEMS can exceed NT (HD=128, NT=128 -> 256 items): loop like the
  // dO staging. QK_IPT==1 for every other variant, so those paths are
  // unchanged. (Single-shot staging here silently dropped half the Q tile
  // for hd-128 small-N — caught by tests/test_kernel_layouts.py.)
  constexpr int QK_IPT = (QK_ITEMS + NT - 1) / NT;
  constexpr int DO_IPT = (DO_ITEMS + NT - 1) / NT;
  bf16x8 qlo[QK_IPT], qhi[QK_IPT], doreg[DO_IPT];
  float lse_reg = INFINITY, d_reg = 0.f;
  auto issue_loads = [&](int qbase0) {
#pragma unroll
    for (int j = 0; j < QK_IPT; ++j) {
      const int idx = threadIdx.x + j * NT;
      const int qrow = qbase0 + idx / PAIRS_PER_ROW;
      const int c0 = (idx % PAIRS_PER_ROW) * 8;
      const bool ok = idx < QK_ITEMS && qrow < N;
      qlo[j] = ok ? load8(qv.at(qrow, 0, c0)) : bf16x8{};
      qhi[j] = ok ? load8(qv.at(qrow, 0, c0 + HALF)) : bf16x8{};
    }
#pragma unroll
    for (int j = 0; j < DO_IPT; ++j) {
      const int idx = threadIdx.x + j * NT;
      const int qrow = qbase0 + idx / PER_ROW;
      doreg[j] = (idx < DO_ITEMS && qrow < N)
                     ? load8(do_base + (long)qrow * do_stride + (idx % PER_ROW) * 8)
                     : bf16x8{};
    }
    if (threadIdx.x < QB) {
      const int qrow = qbase0 + threadIdx.x;
      lse_reg = (qrow < N) ? lse[((long)b * H + h) * N + qrow] : INFINITY;
      d_reg = (qrow < N) ? D[((long)b * H + h) * N + qrow] : 0.f;
    }
  };
  auto write_tile = [&](int qbase0) {
#pragma unroll
    for (int j = 0; j < QK_IPT; ++j) {
      const int idx = threadIdx.x + j * NT;
      if (idx < QK_ITEMS) {
        const int lrow = idx / PAIRS_PER_ROW;
        const int c0 = (idx % PAIRS_PER_ROW) * 8;
        const int qrow = qbase0 + lrow;
        const int p = qrow - prefix;
        if (use_rope && qrow < N && p >= 0)
          rope_rotate8(qlo[j], qhi[j], sin_t + (long)p * HD, cos_t + (long)p * HD, c0);
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          qt_lds[(c0 + e) * QT_STRIDE + lrow] = reinterpret_cast<__hip_bfloat16*>(&qlo[j])[e];
          qt_lds[(c0 + HALF + e) * QT_STRIDE + lrow] = reinterpret_cast<__hip_bfloat16*>(&qhi[j])[e];
        }
      }
    }
#pragma unroll
    for (int j = 0; j < DO_IPT; ++j) {
      const int idx = threadIdx.x + j * NT;
      if (idx < DO_ITEMS) {
        const int lrow = idx / PER_ROW;
        const int c8 = (idx % PER_ROW) * 8;
#pragma unroll
        for (int e = 0; e < 8; ++e)
          dot_lds[(c8 + e) * QT_STRIDE + lrow] = reinterpret_cast<__hip_bfloat16*>(&doreg[j])[e];
      }
    }
    if (threadIdx.x < QB) {
      lse_lds[threadIdx.x] = lse_reg;
      d_lds[threadIdx.x] = d_reg;
    }
  };

  const int n_q = (N + QB - 1) / QB;
  issue_loads(0);
  for (int qt = 0; qt < n_q; ++qt) {
    const int qbase0 = qt * QB;
    __syncthreads();
    write_tile(qbase0);
    if (qt + 1 < n_q) issue_loads(qbase0 + QB);
    __syncthreads();

   {
    const int qbase = qbase0;
    const int qrow_off = 0;
    f32x16 s_acc = {}, dp_acc = {};
#pragma unroll
    for (int s = 0; s < KSLICES; ++s) {
      union { unsigned u[4]; bf16x8 v8; } aq, ad;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const int d = s * 16 + hhalf * 8 + e;
        reinterpret_cast<__hip_bfloat16*>(&aq)[e] = qt_lds[d * QT_STRIDE + qrow_off + l31];
        if constexpr (MODE != 1)
          reinterpret_cast<__hip_bfloat16*>(&ad)[e] = dot_lds[d * QT_STRIDE + qrow_off + l31];
      }
      s_acc = MFMA32(aq.v8, kf[s], s_acc);
      if constexpr (MODE != 1) dp_acc = MFMA32(ad.v8, vf[s], dp_acc);
    }
    float pv[16], ds[16];
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int qrow = qbase + c_row(r, hhalf);
      const bool valid = (qrow < N) && (krow < N);
      const float l = lse_lds[qrow_off + c_row(r, hhalf)];
      float p = valid ? __expf(s_acc[r] * scale - l) : 0.f;
      pv[r] = p;
      if constexpr (MODE != 1)
        ds[r] = p * (dp_acc[r] - d_lds[qrow_off + c_row(r, hhalf)]) * scale;
    }
    bf16x8 p0, p1, s0, s1;
    if constexpr (MODE != 2) {
      p0 = pack_fragment(pv, 0);
      p1 = pack_fragment(pv, 8);
    }
    if constexpr (MODE != 1) {
      s0 = pack_fragment(ds, 0);
      s1 = pack_fragment(ds, 8);
    }
#pragma unroll
    for (int t = 0; t < DTILES; ++t) {
      union { unsigned u[4]; bf16x8 v8; } ado0, ado1, aq0, aq1;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        if constexpr (MODE != 2) {
          reinterpret_cast<__hip_bfloat16*>(&ado0)[e] = dot_lds[(t * 32 + l31) * QT_STRIDE + qrow_off + hhalf * 8 + e];
          reinterpret_cast<__hip_bfloat16*>(&ado1)[e] = dot_lds[(t * 32 + l31) * QT_STRIDE + qrow_off + 16 + hhalf * 8 + e];
        }
        if constexpr (MODE != 1) {
          reinterpret_cast<__hip_bfloat16*>(&aq0)[e] = qt_lds[(t * 32 + l31) * QT_STRIDE + qrow_off + hhalf * 8 + e];
          reinterpret_cast<__hip_bfloat16*>(&aq1)[e] = qt_lds[(t * 32 + l31) * QT_STRIDE + qrow_off + 16 + hhalf * 8 + e];
        }
      }
      if constexpr (MODE != 2) {
        f32x16 accv;
#pragma unroll
        for (int r = 0; r < 16; ++r) accv[r] = dv_acc[t][r];
        accv = MFMA32(ado0.v8, p0, accv);
        accv = MFMA32(ado1.v8, p1, accv);
#pragma unroll
        for (int r = 0; r < 16; ++r) dv_acc[t][r] = accv[r];
      }
      if constexpr (MODE != 1) {
        f32x16 acck;
#pragma unroll
        for (int r = 0; r < 16; ++r) acck[r] = dk_acc[t][r];
        acck = MFMA32(aq0.v8, s0, acck);
        acck = MFMA32(aq1.v8, s1, acck);
#pragma unroll
        for (int r = 0; r < 16; ++r) dk_acc[t][r] = acck[r];
      }
    }
   }
  }

  if (krow < N) {
    const int p = krow - prefix;
    if constexpr (MODE != 1) {
      if (use_rope && p >= 0) {
        const float* srow = sin_t + (long)p * HD;
        const float* crow = cos_t + (long)p * HD;
#pragma unroll
        for (int t = 0; t < DTILES / 2; ++t)
#pragma unroll
          for (int r = 0; r < 16; ++r) {
            const int d = t * 32 + c_row(r, hhalf);
            const float c = crow[d], s = srow[d];
            const float glo = dk_acc[t][r], ghi = dk_acc[t + DTILES / 2][r];
            dk_acc[t][r] = glo * c + ghi * s;
            dk_acc[t + DTILES / 2][r] = ghi * c - glo * s;
          }
      }
    }
    __hip_bfloat16* dkp = dqkv + ((((long)b * N + krow) * 3 + 1) * H + h) * HD;
    __hip_bfloat16* dvp = dqkv + ((((long)b * N + krow) * 3 + 2) * H + h) * HD;
#pragma unroll
    for (int t = 0; t < DTILES; ++t)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int d = t * 32 + c_row(r, hhalf);
        if constexpr (MODE != 1)
          *reinterpret_cast<short*>(dkp + d) = f32_to_bf16(dk_acc[t][r]);
        if constexpr (MODE != 2)
          *reinterpret_cast<short*>(dvp + d) = f32_to_bf16(dv_acc[t][r]);
      }
  }
}

// ---------------------------------------------------------------------------
// Backward dK/dV, 64-row super-tile variant (gated: DINOV3_FMHA_DKV64=1).
//
// Stages TWO 32-row q-tiles per barrier pair (half the __syncthreads and
// lse/D loads of bwd_dkv_kernel). The round-1 attempt of this NaN'd because
// the 64 staged rows kept the 32-row column stride (QB+8=40): rows 40..63 of
// every d-column overwrote the next column, and lse/d overlapped (d_lds was
// offset by QB). Fixed here: QT_STRIDE = 2*QB+8 = 72 and 2*QB-sized lse/d —
// matching the stride the fwd/dq kernels already use. Validate on hardware
// (tests/test_fmha_rope_gpu.py::test_dkv64_variant) before enabling.
// ---------------------------------------------------------------------------
template <int HD, int NT = 256, int MODE = 0>
__global__ __launch_bounds__(NT) void bwd_dkv64_kernel(
    const __hip_bfloat16* __restrict__ qkv, const __hip_bfloat16* __restrict__ dout,
    const float* __restrict__ sin_t, const float* __restrict__ cos_t,
    const float* __restrict__ lse, const float* __restrict__ D,
    __hip_bfloat16* __restrict__ dqkv, int B, int H, int N, int P, float scale) {
  constexpr int KSLICES = HD / 16;
  constexpr int DTILES = HD / 32;
  constexpr int QB = 32;
  constexpr int SQB = 2 * QB;            // staged rows per barrier pair
  constexpr int QT_STRIDE = SQB + 8;     // 72: column stride sized for SQB rows
  constexpr int HALF = HD / 2;

  const int bh = blockIdx.x;
  const int b = bh / H;
  const int h = bh % H;
  const int wave = threadIdx.x / 64;
  const int lane = threadIdx.x & 63;
  const int hhalf = lane >> 5;
  const int l31 = lane & 31;
  const int k0 = blockIdx.y * 128 + wave * 32;
  const int prefix = N - P;
  const bool use_rope = (sin_t != nullptr);

  QkvView qv;
  qv.base = qkv + ((long)b * N * 3 * H + h) * HD;
  qv.row_stride = (long)3 * H * HD;
  qv.which_stride = (long)H * HD;
  const __hip_bfloat16* do_base = dout + ((long)b * N * H + h) * HD;
  const long do_stride = (long)H * HD;

  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  __hip_bfloat16* qt_lds = reinterpret_cast<__hip_bfloat16*>(smem_raw);  // rope'd Q^T [HD][72]
  __hip_bfloat16* dot_lds = qt_lds + HD * QT_STRIDE;                     // dO^T [HD][72]
  float* lse_lds = reinterpret_cast<float*>(dot_lds + HD * QT_STRIDE);   // [SQB]
  float* d_lds = lse_lds + SQB;                                          // [SQB]

  const int krow = k0 + l31;
  const int safe = krow < N ? krow : (N - 1);
  bf16x8 kf[KSLICES], vf[MODE == 1 ? 1 : KSLICES];
  {
#pragma unroll
    for (int s = 0; s < KSLICES; ++s) {
      kf[s] = load8(qv.at(safe, 1, s * 16 + hhalf * 8));
      if constexpr (MODE != 1) vf[s] = load8(qv.at(safe, 2, s * 16 + hhalf * 8));
    }
    const int p = safe - prefix;
    if (use_rope && p >= 0) {
      const float* srow = sin_t + (long)p * HD;
      const float* crow = cos_t + (long)p * HD;
#pragma unroll
      for (int s2 = 0; s2 < KSLICES / 2; ++s2)
        rope_rotate8(kf[s2], kf[s2 + KSLICES / 2], srow, crow, s2 * 16 + hhalf * 8);
    }
  }

  float dk_acc[MODE == 1 ? 1 : DTILES][16], dv_acc[MODE == 2 ? 1 : DTILES][16];
#pragma unroll
  for (int t = 0; t < DTILES; ++t)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      if constexpr (MODE != 1) dk_acc[t][r] = 0.f;
      if constexpr (MODE != 2) dv_acc[t][r] = 0.f;
    }

  constexpr int PAIRS_PER_ROW = HALF / 8;
  constexpr int PER_ROW = HD / 8;
  constexpr int QK_ITEMS = SQB * PAIRS_PER_ROW;  // up to 512 (HD=128)
  constexpr int DO_ITEMS = SQB * PER_ROW;        // up to 1024 (HD=128)
  constexpr int QK_IPT = (QK_ITEMS + NT - 1) / NT;
  constexpr int DO_IPT = (DO_ITEMS + NT - 1) / NT;
  bf16x8 qlo[QK_IPT], qhi[QK_IPT], doreg[DO_IPT];
  float lse_reg = INFINITY, d_reg = 0.f;
  auto issue_loads = [&](int qbase0) {
#pragma unroll
    for (int j = 0; j < QK_IPT; ++j) {
      const int idx = threadIdx.x + j * NT;
      const int qrow = qbase0 + idx / PAIRS_PER_ROW;
      const int c0 = (idx % PAIRS_PER_ROW) * 8;
      const bool ok = idx < QK_ITEMS && qrow < N;
      qlo[j] = ok ? load8(qv.at(qrow, 0, c0)) : bf16x8{};
      qhi[j] = ok ? load8(qv.at(qrow, 0, c0 + HALF)) : bf16x8{};
    }
#pragma unroll
    for (int j = 0; j < DO_IPT; ++j) {
      const int idx = threadIdx.x + j * NT;
      const int qrow = qbase0 + idx / PER_ROW;
      doreg[j] = (idx < DO_ITEMS && qrow < N)
                     ? load8(do_base + (long)qrow * do_stride + (idx % PER_ROW) * 8)
                     : bf16x8{};
    }
    if (threadIdx.x < SQB) {
      const int qrow = qbase0 + threadIdx.x;
      lse_reg = (qrow < N) ? lse[((long)b * H + h) * N + qrow] : INFINITY;
      d_reg = (qrow < N) ? D[((long)b * H + h) * N + qrow] : 0.f;
    }
  };
  auto write_tile = [&](int qbase0) {
#pragma unroll
    for (int j = 0; j < QK_IPT; ++j) {
      const int idx = threadIdx.x + j * NT;
      if (idx < QK_ITEMS) {
        const int lrow = idx / PAIRS_PER_ROW;
        const int c0 = (idx % PAIRS_PER_ROW) * 8;
        const int qrow = qbase0 + lrow;
        const int p = qrow - prefix;
        if (use_rope && qrow < N && p >= 0)
          rope_rotate8(qlo[j], qhi[j], sin_t + (long)p * HD, cos_t + (long)p * HD, c0);
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          qt_lds[(c0 + e) * QT_STRIDE + lrow] = reinterpret_cast<__hip_bfloat16*>(&qlo[j])[e];
          qt_lds[(c0 + HALF + e) * QT_STRIDE + lrow] = reinterpret_cast<__hip_bfloat16*>(&qhi[j])[e];
        }
      }
    }
#pragma unroll
    for (int j = 0; j < DO_IPT; ++j) {
      const int idx = threadIdx.x + j * NT;
      if (idx < DO_ITEMS) {
        const int lrow = idx / PER_ROW;
        const int c8 = (idx % PER_ROW) * 8;
#pragma unroll
        for (int e = 0; e < 8; ++e)
          dot_lds[(c8 + e) * QT_STRIDE + lrow] = reinterpret_cast<__hip_bfloat16*>(&doreg[j])[e];
      }
    }
    if (threadIdx.x < SQB) {
      lse_lds[threadIdx.x] = lse_reg;
      d_lds[threadIdx.x] = d_reg;
    }
  };

  const int n_super = (N + SQB - 1) / SQB;
  issue_loads(0);
  for (int qt = 0; qt < n_super; ++qt) {
    const int qbase0 = qt * SQB;
    __syncthreads();
    write_tile(qbase0);
    if (qt + 1 < n_super) issue_loads(qbase0 + SQB);
    __syncthreads();

    for (int sub = 0; sub < 2 && qbase0 + sub * QB < N; ++sub) {
      const int qbase = qbase0 + sub * QB;
      const int qrow_off = sub * QB;
      f32x16 s_acc = {}, dp_acc = {};
#pragma unroll
      for (int s = 0; s < KSLICES; ++s) {
        union { unsigned u[4]; bf16x8 v8; } aq, ad;
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          const int d = s * 16 + hhalf * 8 + e;
          reinterpret_cast<__hip_bfloat16*>(&aq)[e] = qt_lds[d * QT_STRIDE + qrow_off + l31];
          reinterpret_cast<__hip_bfloat16*>(&ad)[e] = dot_lds[d * QT_STRIDE + qrow_off + l31];
        }
        s_acc = MFMA32(aq.v8, kf[s], s_acc);
        dp_acc = MFMA32(ad.v8, vf[s], dp_acc);
      }
      float pv[16], ds[16];
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int qrow = qbase + c_row(r, hhalf);
        const bool valid = (qrow < N) && (krow < N);
        const float l = lse_lds[qrow_off + c_row(r, hhalf)];
        float p = valid ? __expf(s_acc[r] * scale - l) : 0.f;
        pv[r] = p;
        ds[r] = p * (dp_acc[r] - d_lds[qrow_off + c_row(r, hhalf)]) * scale;
      }
      bf16x8 p0 = pack_fragment(pv, 0);
      bf16x8 p1 = pack_fragment(pv, 8);
      bf16x8 s0 = pack_fragment(ds, 0);
      bf16x8 s1 = pack_fragment(ds, 8);
#pragma unroll
      for (int t = 0; t < DTILES; ++t) {
        union { unsigned u[4]; bf16x8 v8; } ado0, ado1, aq0, aq1;
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          reinterpret_cast<__hip_bfloat16*>(&ado0)[e] = dot_lds[(t * 32 + l31) * QT_STRIDE + qrow_off + hhalf * 8 + e];
          reinterpret_cast<__hip_bfloat16*>(&ado1)[e] = dot_lds[(t * 32 + l31) * QT_STRIDE + qrow_off + 16 + hhalf * 8 + e];
          reinterpret_cast<__hip_bfloat16*>(&aq0)[e] = qt_lds[(t * 32 + l31) * QT_STRIDE + qrow_off + hhalf * 8 + e];
          reinterpret_cast<__hip_bfloat16*>(&aq1)[e] = qt_lds[(t * 32 + l31) * QT_STRIDE + qrow_off + 16 + hhalf * 8 + e];
        }
        f32x16 accv, acck;
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          accv[r] = dv_acc[t][r];
          acck[r] = dk_acc[t][r];
        }
        accv = MFMA32(ado0.v8, p0, accv);
        accv = MFMA32(ado1.v8, p1, accv);
        acck = MFMA32(aq0.v8, s0, acck);
        acck = MFMA32(aq1.v8, s1, acck);
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          dv_acc[t][r] = accv[r];
          dk_acc[t][r] = acck[r];
        }
      }
    }
  }

  if (krow < N) {
    const int p = krow - prefix;
    if constexpr (MODE != 1) {
      if (use_rope && p >= 0) {
        const float* srow = sin_t + (long)p * HD;
        const float* crow = cos_t + (long)p * HD;
#pragma unroll
        for (int t = 0; t < DTILES / 2; ++t)
#pragma unroll
          for (int r = 0; r < 16; ++r) {
            const int d = t * 32 + c_row(r, hhalf);
            const float c = crow[d], s = srow[d];
            const float glo = dk_acc[t][r], ghi = dk_acc[t + DTILES / 2][r];
            dk_acc[t][r] = glo * c + ghi * s;
            dk_acc[t + DTILES / 2][r] = ghi * c - glo * s;
          }
      }
    }
    __hip_bfloat16* dkp = dqkv + ((((long)b * N + krow) * 3 + 1) * H + h) * HD;
    __hip_bfloat16* dvp = dqkv + ((((long)b * N + krow) * 3 + 2) * H + h) * HD;
#pragma unroll
    for (int t = 0; t < DTILES; ++t)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int d = t * 32 + c_row(r, hhalf);
        if constexpr (MODE != 1)
          *reinterpret_cast<short*>(dkp + d) = f32_to_bf16(dk_acc[t][r]);
        if constexpr (MODE != 2)
          *reinterpret_cast<short*>(dvp + d) = f32_to_bf16(dv_acc[t][r]);
      }
  }
}

// preprocess on token-major dO/O: D[b,h,n] = sum_d dO*O
__global__ void bwd_pre_tm_kernel(const __hip_bfloat16* __restrict__ dout,
                                  const __hip_bfloat16* __restrict__ o,
                                  float* __restrict__ D, int B, int H, int N, int HD) {
  // rows ordered (b, h, n) in D; data token-major (b, n, h, d)
  const long rows = (long)B * H * N;
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x / 64;
  const int sub = lane / 8;
  const int lane8 = lane & 7;
  const long group0 = ((long)blockIdx.x * (blockDim.x / 64) + wid) * 8;
  for (long g = group0; g < rows; g += (long)gridDim.x * (blockDim.x / 64) * 8) {
    const long row = g + sub;
    float acc = 0.f;
    if (row < rows) {
      const int n = (int)(row % N);
      const int h = (int)((row / N) % H);
      const int b = (int)(row / ((long)N * H));
      const long off = (((long)b * N + n) * H + h) * HD;
      for (int i0 = lane8 * 8; i0 < HD; i0 += 64) {
        __hip_bfloat16 a[8], c[8];
        Vec8<__hip_bfloat16>::load(a, dout + off + i0);
        Vec8<__hip_bfloat16>::load(c, o + off + i0);
#pragma unroll
        for (int e = 0; e < 8; ++e)
          acc += bf16_to_f32(*(short*)(a + e)) * bf16_to_f32(*(short*)(c + e));
      }
    }
#pragma unroll
    for (int off = 4; off > 0; off >>= 1) acc += __shfl_down(acc, off, 64);
    if (lane8 == 0 && row < rows) D[row] = acc;
  }
}

}  // namespace fmha_rope

// ---------------------------------------------------------------------------
// launchers
// ---------------------------------------------------------------------------

void launch_fmha_rope_fwd(const __hip_bfloat16* qkv, const float* sin_t,
                          const float* cos_t, __hip_bfloat16* o, float* lse, int B, int H,
                          int N, int P, int HD, float scale, hipStream_t stream) {
  // short sequences (local crops, N <= 64): 128-thread blocks, 64 q rows each
  const bool small = N <= 64;
  dim3 grid(B * H, small ? 1 : (N + 127) / 128);
  if (HD == 64) {
    if (small) {
      size_t shmem = (64 * 72 + 64 * 72) * sizeof(__hip_bfloat16);
      hipLaunchKernelGGL(HIP_KERNEL_NAME(fmha_rope::fwd_kernel<64, 128>), grid, dim3(128),
                         shmem, stream, qkv, sin_t, cos_t, o, lse, B, H, N, P, scale);
    } else {
      // KVB=64: 128 keys per barrier pair; k_lds [128][72], vt_lds [64][136]
      size_t shmem = (128 * 72 + 64 * 136) * sizeof(__hip_bfloat16);
      hipLaunchKernelGGL(
          HIP_KERNEL_NAME(fmha_rope::fwd_kernel<64, 256, false, true, 1, 64>), grid,
          dim3(256), shmem, stream, qkv, sin_t, cos_t, o, lse, B, H, N, P, scale);
    }
  } else if (HD == 128) {
    size_t shmem = (64 * 136 + 128 * 72) * sizeof(__hip_bfloat16);
    if (small) {
      hipLaunchKernelGGL(HIP_KERNEL_NAME(fmha_rope::fwd_kernel<128, 128>), grid, dim3(128),
                         shmem, stream, qkv, sin_t, cos_t, o, lse, B, H, N, P, scale);
    } else {
      // QLDS + no-prefetch: 2 waves/SIMD (the register variant runs at 1)
      auto* kfn = reinterpret_cast<const void*>(
          HIP_KERNEL_NAME(fmha_rope::fwd_kernel<128, 256, true, false, 2>));
      static bool attr_ok = [kfn] {
        return hipFuncSetAttribute(kfn, hipFuncAttributeMaxDynamicSharedMemorySize,
                                   160 * 1024) == hipSuccess;
      }();
      (void)attr_ok;
      size_t shmem_q = shmem + (size_t)(256 / 64) * (128 / 16) * 64 * 8 *
                                   sizeof(__hip_bfloat16);
      hipLaunchKernelGGL(HIP_KERNEL_NAME(fmha_rope::fwd_kernel<128, 256, true, false, 2>),
                         grid, dim3(256), shmem_q, stream, qkv, sin_t, cos_t, o, lse, B,
                         H, N, P, scale);
    }
  }
}

void launch_fmha_rope_bwd_pre(const __hip_bfloat16* dout, const __hip_bfloat16* o,
                              float* D, int B, int H, int N, int HD, hipStream_t stream) {
  long rows = (long)B * H * N;
  int grid = (int)min((rows + 31) / 32, (long)4096);
  hipLaunchKernelGGL(fmha_rope::bwd_pre_tm_kernel, dim3(grid), dim3(256), 0, stream, dout,
                     o, D, B, H, N, HD);
}

void launch_fmha_rope_bwd_dq(const __hip_bfloat16* qkv, const __hip_bfloat16* dout,
                             const float* sin_t, const float* cos_t, const float* lse,
                             const float* D, __hip_bfloat16* dqkv, int B, int H, int N,
                             int P, int HD, float scale, hipStream_t stream) {
  const bool small = N <= 64;
  dim3 grid(B * H, small ? 1 : (N + 127) / 128);
  if (HD == 64) {
    size_t shmem = (2 * 64 * 72 + 64 * 72) * sizeof(__hip_bfloat16);
    if (small)
      hipLaunchKernelGGL(HIP_KERNEL_NAME(fmha_rope::bwd_dq_kernel<64, 128>), grid,
                         dim3(128), shmem, stream, qkv, dout, sin_t, cos_t, lse, D, dqkv,
                         B, H, N, P, scale);
    else {
      // KVB=64: 128 keys per barrier pair (k_lds+v_lds [128][72], kt [64][136])
      size_t shmem64 = (2 * 128 * 72 + 64 * 136) * sizeof(__hip_bfloat16);
      hipLaunchKernelGGL(HIP_KERNEL_NAME(fmha_rope::bwd_dq_kernel<64, 256, 2, 64>), grid,
                         dim3(256), shmem64, stream, qkv, dout, sin_t, cos_t, lse, D,
                         dqkv, B, H, N, P, scale);
    }
  } else if (HD == 128) {
    size_t shmem = (2 * 64 * 136 + 128 * 72) * sizeof(__hip_bfloat16);
    if (small)
      hipLaunchKernelGGL(HIP_KERNEL_NAME(fmha_rope::bwd_dq_kernel<128, 128>), grid,
                         dim3(128), shmem, stream, qkv, dout, sin_t, cos_t, lse, D, dqkv,
                         B, H, N, P, scale);
    else
      // MINW=2 packs to 256 regs (20 B/lane cold scratch) for 2 waves/SIMD
      hipLaunchKernelGGL(HIP_KERNEL_NAME(fmha_rope::bwd_dq_kernel<128, 256, 2>), grid,
                         dim3(256), shmem, stream, qkv, dout, sin_t, cos_t, lse, D, dqkv,
                         B, H, N, P, scale);
  }
}

void launch_fmha_rope_bwd_dkv(const __hip_bfloat16* qkv, const __hip_bfloat16* dout,
                              const float* sin_t, const float* cos_t, const float* lse,
                              const float* D, __hip_bfloat16* dqkv, int B, int H, int N,
                              int P, int HD, float scale, hipStream_t stream) {
  const bool small = N <= 64;
  // Gated 64-row super-tile variant (see bwd_dkv64_kernel); only useful when
  // there is more than one 32-row q-tile. getenv per launch is ~ns vs the
  // 100 us kernel and keeps the flag flippable from tests. Flip
  // DKV64_DEFAULT to 1 in round 2 after hardware validation.
  constexpr bool DKV64_DEFAULT = false;
  const char* dkv64 = getenv("DINOV3_FMHA_DKV64");
  const bool dkv64_on = dkv64 ? (dkv64[0] == '1') : DKV64_DEFAULT;
  const bool use64 = !small && dkv64_on;
  dim3 grid(B * H, small ? 1 : (N + 127) / 128);
  if (HD == 64) {
    size_t shmem = 2 * 64 * 72 * sizeof(__hip_bfloat16) + 128 * sizeof(float);
    if (small)
      hipLaunchKernelGGL(HIP_KERNEL_NAME(fmha_rope::bwd_dkv_kernel<64, 128>), grid,
                         dim3(128), shmem, stream, qkv, dout, sin_t, cos_t, lse, D, dqkv,
                         B, H, N, P, scale);
    else if (use64)
      hipLaunchKernelGGL(HIP_KERNEL_NAME(fmha_rope::bwd_dkv64_kernel<64, 256>), grid,
                         dim3(256), shmem, stream, qkv, dout, sin_t, cos_t, lse, D, dqkv,
                         B, H, N, P, scale);
    else
      hipLaunchKernelGGL(HIP_KERNEL_NAME(fmha_rope::bwd_dkv_kernel<64, 256>), grid,
                         dim3(256), shmem, stream, qkv, dout, sin_t, cos_t, lse, D, dqkv,
                         B, H, N, P, scale);
  } else if (HD == 128) {
    size_t shmem = 2 * 128 * 72 * sizeof(__hip_bfloat16) + 128 * sizeof(float);
    if (small)
      hipLaunchKernelGGL(HIP_KERNEL_NAME(fmha_rope::bwd_dkv_kernel<128, 128>), grid,
                         dim3(128), shmem, stream, qkv, dout, sin_t, cos_t, lse, D, dqkv,
                         B, H, N, P, scale);
    else if (use64)
      hipLaunchKernelGGL(HIP_KERNEL_NAME(fmha_rope::bwd_dkv64_kernel<128, 256>), grid,
                         dim3(256), shmem, stream, qkv, dout, sin_t, cos_t, lse, D, dqkv,
                         B, H, N, P, scale);
    else
      // combined kernel (occupancy 1 from the 366-reg dual accumulators):
      // the MODE=1/2 dV/dK split reaches 2 waves/SIMD but re-streams Q/dO
      // and measured ~5% SLOWER end to end (r2_gpu15) — not used
      hipLaunchKernelGGL(HIP_KERNEL_NAME(fmha_rope::bwd_dkv_kernel<128, 256>), grid,
                         dim3(256), shmem, stream, qkv, dout, sin_t, cos_t, lse, D, dqkv,
                         B, H, N, P, scale);
  }
}
