// Fused multi-head attention for DINOv3 sequence lengths.
//
// Design (MI355X / CDNA4):
// - v_mfma_f32_32x32x16_bf16 tiles; wave64; 4-wave (256-thread) workgroups.
// - forward: one workgroup per (batch*head, 128-row q block); each wave owns
//   32 q rows. K/V tiles of 32 keys staged in LDS (V transposed at staging
//   time so the PV A-fragment reads contiguously). Scores are computed
//   SWAPPED — S^T = K @ Q^T — so each lane holds a full q column and the
//   online-softmax row reduction is 16 lane-local values + one shfl_xor(32).
// - P^T (fp32 accum) repacks into the next MFMA's bf16 fragment with
//   v_cvt_pk_bf16_f32 pairs + v_permlane32_swap (T12/T21 idiom).
// - backward: FA2-style recompute from the saved logsumexp. Three kernels:
//   preprocess D = rowsum(dO*O); dQ (loop over k tiles); dK/dV (loop over q
//   tiles). Same swapped-MFMA + permlane repack machinery.
//
// MFMA fragment layouts assumed (verified on-device by probe_mfma_layout):
//   A[i][k]: lane l holds A[i=l&31][k=(l>>5)*8 + idx], idx=0..7
//   B[k][j]: lane l holds B[k=(l>>5)*8 + idx][j=l&31]
//   C[r][c]: lane l reg r holds C[row=(r&3)+8*(r>>2)+4*(l>>5)][col=l&31]

#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(16))) float f32x16;
typedef __attribute__((ext_vector_type(2))) unsigned int uint2_t;

#define MFMA32(a, b, c) __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0)

DEV_INLINE unsigned cvt_pk_bf16(float lo, float hi) {
  unsigned r;
  asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(lo), "v"(hi));
  return r;
}

// exchange: lanes>=32 of a swap with lanes<32 of b
DEV_INLINE void permlane32_swap(unsigned& a, unsigned& b) {
  auto r = __builtin_amdgcn_permlane32_swap(a, b, false, false);
  a = r[0];
  b = r[1];
}

// Pack 8 fp32 accumulator regs (reg q..q+7 of an S^T C-tile) into one MFMA
// bf16 operand fragment for the NEXT mfma whose k-dim runs over this tile's
// rows. After the swap, (r01, r23) holds k=(h*8)+0..7 for this lane.
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

DEV_INLINE bf16x8 load_bf16x8(const __hip_bfloat16* p) {
  return *reinterpret_cast<const bf16x8*>(p);
}

// row index of C reg r for this lane's half h: (r&3) + 8*(r>>2) + 4*h
DEV_INLINE int c_row(int r, int h) { return (r & 3) + 8 * (r >> 2) + 4 * h; }

// ---------------------------------------------------------------------------
// MFMA layout probe: C[32,32] = A[32,16] @ B[16,32] with the assumed layouts.
// ---------------------------------------------------------------------------
__global__ void probe_mfma_kernel(const __hip_bfloat16* __restrict__ A,
                                  const __hip_bfloat16* __restrict__ B,
                                  float* __restrict__ C) {
  const int lane = threadIdx.x & 63;
  const int h = lane >> 5;
  const int i = lane & 31;
  union {
    unsigned u[4];
    bf16x8 v;
  } a, b;
  for (int x = 0; x < 8; ++x) {
    reinterpret_cast<__hip_bfloat16*>(&a)[x] = A[i * 16 + (h * 8 + x)];
    reinterpret_cast<__hip_bfloat16*>(&b)[x] = B[(h * 8 + x) * 32 + i];
  }
  f32x16 c = {};
  c = MFMA32(a.v, b.v, c);
  for (int r = 0; r < 16; ++r) C[c_row(r, h) * 32 + i] = c[r];
}

// ---------------------------------------------------------------------------
// Forward
// ---------------------------------------------------------------------------
// q,k,v: [BH, N, HD] bf16 contiguous. o: same. lse: [BH, N] fp32.
// Workgroup: 4 waves x 32 q rows = 128-row q block. KV tiles of 32.

template <int HD>
__global__ __launch_bounds__(256) void fmha_fwd_kernel(
    const __hip_bfloat16* __restrict__ q, const __hip_bfloat16* __restrict__ k,
    const __hip_bfloat16* __restrict__ v, __hip_bfloat16* __restrict__ o,
    float* __restrict__ lse, int N, float scale) {
  constexpr int KSLICES = HD / 16;   // MFMA K-steps over head dim
  constexpr int DTILES = HD / 32;    // output d tiles
  constexpr int KVB = 32;            // kv tile
  constexpr int LDS_STRIDE = HD + 8; // +16B pad vs bank conflicts

  const int bh = blockIdx.x;
  const int qblock = blockIdx.y;
  const int wave = threadIdx.x / 64;
  const int lane = threadIdx.x & 63;
  const int hhalf = lane >> 5;
  const int l31 = lane & 31;

  const int q0 = qblock * 128 + wave * 32;  // this wave's first q row
  const long base = (long)bh * N * HD;

  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  __hip_bfloat16* k_lds = reinterpret_cast<__hip_bfloat16*>(smem_raw);           // [KVB][LDS_STRIDE]
  __hip_bfloat16* vt_lds = k_lds + KVB * LDS_STRIDE;                             // [HD][KVB+8]
  constexpr int VT_STRIDE = KVB + 8;

  // --- Q fragments: lane holds Q[q0 + l31][(h*8 + s*16) .. +8) per slice ---
  bf16x8 qf[KSLICES];
  {
    const int qrow = q0 + l31;
    const int safe_row = qrow < N ? qrow : (N - 1);
    const __hip_bfloat16* qp = q + base + (long)safe_row * HD;
#pragma unroll
    for (int s = 0; s < KSLICES; ++s) qf[s] = load_bf16x8(qp + s * 16 + hhalf * 8);
  }

  float o_acc[DTILES][16];
#pragma unroll
  for (int t = 0; t < DTILES; ++t)
#pragma unroll
    for (int r = 0; r < 16; ++r) o_acc[t][r] = 0.f;
  float m_run = -INFINITY;
  float l_run = 0.f;

  const int n_kv_tiles = (N + KVB - 1) / KVB;
  for (int kt = 0; kt < n_kv_tiles; ++kt) {
    const int kbase = kt * KVB;
    // --- stage K tile [KVB][HD] and V^T tile [HD][KVB] ---
    __syncthreads();
    {
      // 256 threads; K: each thread loads 8 bf16 (KVB*HD/8 = 256 for HD=64)
      constexpr int PER_ROW = HD / 8;  // threads per row
      for (int idx = threadIdx.x; idx < KVB * PER_ROW; idx += 256) {
        const int row = idx / PER_ROW;
        const int col8 = (idx % PER_ROW) * 8;
        const int krow = kbase + row;
        bf16x8 kv;
        if (krow < N) {
          kv = load_bf16x8(k + base + (long)krow * HD + col8);
        } else {
          kv = bf16x8{};
        }
        *reinterpret_cast<bf16x8*>(&k_lds[row * LDS_STRIDE + col8]) = kv;
        // V transposed staging (scalar scatter; optimize later with tr-reads)
        bf16x8 vv;
        if (krow < N) {
          vv = load_bf16x8(v + base + (long)krow * HD + col8);
        } else {
          vv = bf16x8{};
        }
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          vt_lds[(col8 + e) * VT_STRIDE + row] = reinterpret_cast<__hip_bfloat16*>(&vv)[e];
        }
      }
    }
    __syncthreads();

    // --- S^T tile = K @ Q^T : C[k row, q col], k rows = this kv tile ---
    f32x16 s_acc = {};
#pragma unroll
    for (int s = 0; s < KSLICES; ++s) {
      bf16x8 af = load_bf16x8(&k_lds[l31 * LDS_STRIDE + s * 16 + hhalf * 8]);
      s_acc = MFMA32(af, qf[s], s_acc);
    }
    float sv[16];
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int krow = kbase + c_row(r, hhalf);
      sv[r] = (krow < N) ? s_acc[r] * scale : -INFINITY;
    }
    // --- online softmax: row (q) stats over this tile's 32 k ---
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

    // --- PV: O^T[d, q] += V^T[d, k] @ P^T[k, q] ---
    bf16x8 pfrag0 = pack_fragment(sv, 0);
    bf16x8 pfrag1 = pack_fragment(sv, 8);
#pragma unroll
    for (int t = 0; t < DTILES; ++t) {
      // A = V^T rows d = t*32 + l31, k = h*8.. (slice 0: k 0-15, slice 1: 16-31)
      bf16x8 a0 = load_bf16x8(&vt_lds[(t * 32 + l31) * VT_STRIDE + hhalf * 8]);
      bf16x8 a1 = load_bf16x8(&vt_lds[(t * 32 + l31) * VT_STRIDE + 16 + hhalf * 8]);
      f32x16 acc;
#pragma unroll
      for (int r = 0; r < 16; ++r) acc[r] = o_acc[t][r];
      acc = MFMA32(a0, pfrag0, acc);
      acc = MFMA32(a1, pfrag1, acc);
#pragma unroll
      for (int r = 0; r < 16; ++r) o_acc[t][r] = acc[r];
    }
  }

  // --- epilogue: O[q, d] = O^T / l ; lse = m + log(l) ---
  const float inv_l = 1.0f / l_run;
  const int qrow = q0 + l31;
  if (qrow < N) {
    __hip_bfloat16* op = o + base + (long)qrow * HD;
#pragma unroll
    for (int t = 0; t < DTILES; ++t)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int d = t * 32 + c_row(r, hhalf);
        *reinterpret_cast<short*>(op + d) = f32_to_bf16(o_acc[t][r] * inv_l);
      }
    if (hhalf == 0) lse[(long)bh * N + qrow] = m_run + __logf(l_run);
  }
}

// ---------------------------------------------------------------------------
// Backward preprocess: D[bh, n] = sum_d dO[bh,n,d] * O[bh,n,d]  (fp32)
// ---------------------------------------------------------------------------
__global__ void fmha_bwd_pre_kernel(const __hip_bfloat16* __restrict__ dout,
                                    const __hip_bfloat16* __restrict__ o,
                                    float* __restrict__ D, long rows, int HD) {
  // one wave per 8 rows: 8 lanes per row, 8 bf16 per lane per pass
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x / 64;
  const int sub = lane / 8;           // row within the wave's 8-row group
  const int lane8 = lane & 7;
  const long group0 = ((long)blockIdx.x * (blockDim.x / 64) + wid) * 8;
  for (long g = group0; g < rows; g += (long)gridDim.x * (blockDim.x / 64) * 8) {
    const long row = g + sub;
    float acc = 0.f;
    if (row < rows) {
      const __hip_bfloat16* dop = dout + row * HD;
      const __hip_bfloat16* op = o + row * HD;
      for (int i0 = lane8 * 8; i0 < HD; i0 += 64) {
        __hip_bfloat16 a[8], b[8];
        Vec8<__hip_bfloat16>::load(a, dop + i0);
        Vec8<__hip_bfloat16>::load(b, op + i0);
#pragma unroll
        for (int e = 0; e < 8; ++e)
          acc += bf16_to_f32(*(short*)(a + e)) * bf16_to_f32(*(short*)(b + e));
      }
    }
    // reduce across the 8 lanes of this row
#pragma unroll
    for (int off = 4; off > 0; off >>= 1) acc += __shfl_down(acc, off, 64);
    if (lane8 == 0 && row < rows) D[row] = acc;
  }
}

// ---------------------------------------------------------------------------
// Backward dQ: per q block, loop over kv tiles.
//   S^T = K @ Q^T (recompute) ; P^T = exp(S^T*scale - lse[q])
//   dP^T[k,q] = V @ dO^T       ; dS^T = P^T * (dP^T - D[q]) * scale
//   dQ^T[d,q] += K^T[d,k-slice] @ pack(dS^T)
// ---------------------------------------------------------------------------
template <int HD>
__global__ __launch_bounds__(256) void fmha_bwd_dq_kernel(
    const __hip_bfloat16* __restrict__ q, const __hip_bfloat16* __restrict__ k,
    const __hip_bfloat16* __restrict__ v, const __hip_bfloat16* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ D,
    __hip_bfloat16* __restrict__ dq, int N, float scale) {
  constexpr int KSLICES = HD / 16;
  constexpr int DTILES = HD / 32;
  constexpr int KVB = 32;
  constexpr int LDS_STRIDE = HD + 8;
  constexpr int KT_STRIDE = KVB + 8;

  const int bh = blockIdx.x;
  const int qblock = blockIdx.y;
  const int wave = threadIdx.x / 64;
  const int lane = threadIdx.x & 63;
  const int hhalf = lane >> 5;
  const int l31 = lane & 31;
  const int q0 = qblock * 128 + wave * 32;
  const long base = (long)bh * N * HD;

  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  __hip_bfloat16* k_lds = reinterpret_cast<__hip_bfloat16*>(smem_raw);  // [KVB][LDS_STRIDE]
  __hip_bfloat16* v_lds = k_lds + KVB * LDS_STRIDE;                      // [KVB][LDS_STRIDE]
  __hip_bfloat16* kt_lds = v_lds + KVB * LDS_STRIDE;                     // [HD][KVB+8]

  const int qrow = q0 + l31;
  const int safe_row = qrow < N ? qrow : (N - 1);
  bf16x8 qf[KSLICES], dof[KSLICES];
  {
    const __hip_bfloat16* qp = q + base + (long)safe_row * HD;
    const __hip_bfloat16* dop = dout + base + (long)safe_row * HD;
#pragma unroll
    for (int s = 0; s < KSLICES; ++s) {
      qf[s] = load_bf16x8(qp + s * 16 + hhalf * 8);
      dof[s] = load_bf16x8(dop + s * 16 + hhalf * 8);
    }
  }
  const float my_lse = lse[(long)bh * N + safe_row];
  const float my_D = D[(long)bh * N + safe_row];

  float dq_acc[DTILES][16];
#pragma unroll
  for (int t = 0; t < DTILES; ++t)
#pragma unroll
    for (int r = 0; r < 16; ++r) dq_acc[t][r] = 0.f;

  const int n_kv_tiles = (N + KVB - 1) / KVB;
  for (int kt = 0; kt < n_kv_tiles; ++kt) {
    const int kbase = kt * KVB;
    __syncthreads();
    {
      constexpr int PER_ROW = HD / 8;
      for (int idx = threadIdx.x; idx < KVB * PER_ROW; idx += 256) {
        const int row = idx / PER_ROW;
        const int col8 = (idx % PER_ROW) * 8;
        const int krow = kbase + row;
        bf16x8 kv = (krow < N) ? load_bf16x8(k + base + (long)krow * HD + col8) : bf16x8{};
        bf16x8 vv = (krow < N) ? load_bf16x8(v + base + (long)krow * HD + col8) : bf16x8{};
        *reinterpret_cast<bf16x8*>(&k_lds[row * LDS_STRIDE + col8]) = kv;
        *reinterpret_cast<bf16x8*>(&v_lds[row * LDS_STRIDE + col8]) = vv;
#pragma unroll
        for (int e = 0; e < 8; ++e)
          kt_lds[(col8 + e) * KT_STRIDE + row] = reinterpret_cast<__hip_bfloat16*>(&kv)[e];
      }
    }
    __syncthreads();

    f32x16 s_acc = {}, dp_acc = {};
#pragma unroll
    for (int s = 0; s < KSLICES; ++s) {
      bf16x8 kf = load_bf16x8(&k_lds[l31 * LDS_STRIDE + s * 16 + hhalf * 8]);
      bf16x8 vf = load_bf16x8(&v_lds[l31 * LDS_STRIDE + s * 16 + hhalf * 8]);
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
      bf16x8 a0 = load_bf16x8(&kt_lds[(t * 32 + l31) * KT_STRIDE + hhalf * 8]);
      bf16x8 a1 = load_bf16x8(&kt_lds[(t * 32 + l31) * KT_STRIDE + 16 + hhalf * 8]);
      f32x16 acc;
#pragma unroll
      for (int r = 0; r < 16; ++r) acc[r] = dq_acc[t][r];
      acc = MFMA32(a0, f0, acc);
      acc = MFMA32(a1, f1, acc);
#pragma unroll
      for (int r = 0; r < 16; ++r) dq_acc[t][r] = acc[r];
    }
  }

  if (qrow < N) {
    __hip_bfloat16* dqp = dq + base + (long)qrow * HD;
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
// Backward dK/dV: per kv block, loop over q tiles.
//   S[q,k] = Q @ K^T (C: rows=q, cols=k; K wave-resident as B)
//   P = exp(S*scale - lse[q]); dP[q,k] = dO @ V^T
//   dS = P*(dP - D[q])*scale
//   dV[k,d] += pack(P)^T-as-A @ dO(from LDS as B)
//   dK[k,d] += pack(dS) @ Q^T(from LDS as B)
// ---------------------------------------------------------------------------
template <int HD>
__global__ __launch_bounds__(256) void fmha_bwd_dkv_kernel(
    const __hip_bfloat16* __restrict__ q, const __hip_bfloat16* __restrict__ k,
    const __hip_bfloat16* __restrict__ v, const __hip_bfloat16* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ D,
    __hip_bfloat16* __restrict__ dk, __hip_bfloat16* __restrict__ dv, int N, float scale) {
  constexpr int KSLICES = HD / 16;
  constexpr int DTILES = HD / 32;
  constexpr int QB = 32;
  constexpr int QT_STRIDE = QB + 8;

  const int bh = blockIdx.x;
  const int kblock = blockIdx.y;
  const int wave = threadIdx.x / 64;
  const int lane = threadIdx.x & 63;
  const int hhalf = lane >> 5;
  const int l31 = lane & 31;
  const int k0 = kblock * 128 + wave * 32;  // this wave's first k row
  const long base = (long)bh * N * HD;

  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  // Q^T and dO^T tiles: [HD][QB+8] each; plus lse/D tiles [QB]
  __hip_bfloat16* qt_lds = reinterpret_cast<__hip_bfloat16*>(smem_raw);
  __hip_bfloat16* dot_lds = qt_lds + HD * QT_STRIDE;
  float* lse_lds = reinterpret_cast<float*>(dot_lds + HD * QT_STRIDE);
  float* d_lds = lse_lds + QB;

  const int krow = k0 + l31;
  const int safe_k = krow < N ? krow : (N - 1);
  bf16x8 kf[KSLICES], vf[KSLICES];
  {
    const __hip_bfloat16* kp = k + base + (long)safe_k * HD;
    const __hip_bfloat16* vp = v + base + (long)safe_k * HD;
#pragma unroll
    for (int s = 0; s < KSLICES; ++s) {
      kf[s] = load_bf16x8(kp + s * 16 + hhalf * 8);
      vf[s] = load_bf16x8(vp + s * 16 + hhalf * 8);
    }
  }

  float dk_acc[DTILES][16], dv_acc[DTILES][16];
#pragma unroll
  for (int t = 0; t < DTILES; ++t)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      dk_acc[t][r] = 0.f;
      dv_acc[t][r] = 0.f;
    }

  const int n_q_tiles = (N + QB - 1) / QB;
  for (int qt = 0; qt < n_q_tiles; ++qt) {
    const int qbase = qt * QB;
    __syncthreads();
    {
      constexpr int PER_ROW = HD / 8;
      for (int idx = threadIdx.x; idx < QB * PER_ROW; idx += 256) {
        const int row = idx / PER_ROW;
        const int col8 = (idx % PER_ROW) * 8;
        const int qrow = qbase + row;
        bf16x8 qv = (qrow < N) ? load_bf16x8(q + base + (long)qrow * HD + col8) : bf16x8{};
        bf16x8 dov = (qrow < N) ? load_bf16x8(dout + base + (long)qrow * HD + col8) : bf16x8{};
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          qt_lds[(col8 + e) * QT_STRIDE + row] = reinterpret_cast<__hip_bfloat16*>(&qv)[e];
          dot_lds[(col8 + e) * QT_STRIDE + row] = reinterpret_cast<__hip_bfloat16*>(&dov)[e];
        }
      }
      for (int row = threadIdx.x; row < QB; row += 256) {
        const int qrow = qbase + row;
        lse_lds[row] = (qrow < N) ? lse[(long)bh * N + qrow] : INFINITY;
        d_lds[row] = (qrow < N) ? D[(long)bh * N + qrow] : 0.f;
      }
    }
    __syncthreads();

    // S[q,k]: A = Q tile rows q (from qt_lds, transposed back: A[i=q][d])...
    // A fragment needs A[i=l31][d=h*8+idx] = Q[qbase+l31][d] — qt_lds holds
    // Q^T[d][q]: read 8 strided bf16. Instead use B-side trick:
    //   S^T2[q,k]? We want C rows=q, cols=k with K as B-fragment:
    //   B[d][j=k=l31] = K[k][d] -> kf (wave-resident) ✓ is A-layout not B.
    // kf was loaded as lane l31 -> K[k=l31][d=h*8+idx]: that IS the B layout
    // B[k-dim=d][j=l31] for matrix K^T (d rows, k cols). So:
    //   C = A(Q) @ B(K^T) with A[i=q][d] taken from qt_lds via strided reads.
    f32x16 s_acc = {}, dp_acc = {};
#pragma unroll
    for (int s = 0; s < KSLICES; ++s) {
      union {
        unsigned u[4];
        bf16x8 v8;
      } aq, ad;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const int d = s * 16 + hhalf * 8 + e;
        reinterpret_cast<__hip_bfloat16*>(&aq)[e] = qt_lds[d * QT_STRIDE + l31];
        reinterpret_cast<__hip_bfloat16*>(&ad)[e] = dot_lds[d * QT_STRIDE + l31];
      }
      s_acc = MFMA32(aq.v8, kf[s], s_acc);
      dp_acc = MFMA32(ad.v8, vf[s], dp_acc);
    }
    float pv[16], ds[16];
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int qrow = qbase + c_row(r, hhalf);
      const bool valid = (qrow < N) && (krow < N);
      const float l = lse_lds[c_row(r, hhalf)];
      float p = valid ? __expf(s_acc[r] * scale - l) : 0.f;
      pv[r] = p;
      ds[r] = p * (dp_acc[r] - d_lds[c_row(r, hhalf)]) * scale;
    }
    bf16x8 p0 = pack_fragment(pv, 0);
    bf16x8 p1 = pack_fragment(pv, 8);
    bf16x8 s0 = pack_fragment(ds, 0);
    bf16x8 s1 = pack_fragment(ds, 8);
    // dV[k, d] += P^T-as-A @ dO-as-B ; dK[k, d] += dS^T-as-A @ Q-as-B
    // A fragment: pack gives lane -> [k-dim = q slice][i = l31] — but C rows
    // must be k. Wait: pack_fragment of C-regs (rows=q, cols=k) yields for
    // lane j=l31(=k col) the q values (h*8+idx): that is layout
    // X[q=(h*8+idx)][k=l31] == B-layout of matrix P (q rows, k cols)!
    // So use it as B, and take dO^T / Q^T as A (from the transposed LDS
    // tiles): C[d, k] = dO^T[d, q] @ P[q, k]  -> rows=d, cols=k.
#pragma unroll
    for (int t = 0; t < DTILES; ++t) {
      // dO^T / Q^T A fragments: A[i = d = t*32+l31][q-dim = h*8+idx]
      union {
        unsigned u[4];
        bf16x8 v8;
      } ado0, ado1, aq0, aq1;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        reinterpret_cast<__hip_bfloat16*>(&ado0)[e] = dot_lds[(t * 32 + l31) * QT_STRIDE + hhalf * 8 + e];
        reinterpret_cast<__hip_bfloat16*>(&ado1)[e] = dot_lds[(t * 32 + l31) * QT_STRIDE + 16 + hhalf * 8 + e];
        reinterpret_cast<__hip_bfloat16*>(&aq0)[e] = qt_lds[(t * 32 + l31) * QT_STRIDE + hhalf * 8 + e];
        reinterpret_cast<__hip_bfloat16*>(&aq1)[e] = qt_lds[(t * 32 + l31) * QT_STRIDE + 16 + hhalf * 8 + e];
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

  // dK/dV accumulators are C[d rows, k cols]: lane col = k = l31 ✓ but rows
  // are d — transposed store like the fwd epilogue.
  if (krow < N) {
    __hip_bfloat16* dkp = dk + base + (long)krow * HD;
    __hip_bfloat16* dvp = dv + base + (long)krow * HD;
#pragma unroll
    for (int t = 0; t < DTILES; ++t)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int d = t * 32 + c_row(r, hhalf);
        *reinterpret_cast<short*>(dkp + d) = f32_to_bf16(dk_acc[t][r]);
        *reinterpret_cast<short*>(dvp + d) = f32_to_bf16(dv_acc[t][r]);
      }
  }
}

// ---------------------------------------------------------------------------
// launchers
// ---------------------------------------------------------------------------

void launch_probe_mfma(const __hip_bfloat16* A, const __hip_bfloat16* B, float* C,
                       hipStream_t stream) {
  hipLaunchKernelGGL(probe_mfma_kernel, dim3(1), dim3(64), 0, stream, A, B, C);
}

void launch_fmha_fwd(const __hip_bfloat16* q, const __hip_bfloat16* k,
                     const __hip_bfloat16* v, __hip_bfloat16* o, float* lse, int BH,
                     int N, int HD, float scale, hipStream_t stream) {
  dim3 grid(BH, (N + 127) / 128);
  if (HD == 64) {
    size_t shmem = (32 * (64 + 8) + 64 * (32 + 8)) * sizeof(__hip_bfloat16);
    hipLaunchKernelGGL((fmha_fwd_kernel<64>), grid, dim3(256), shmem, stream, q, k, v, o,
                       lse, N, scale);
  } else if (HD == 128) {
    size_t shmem = (32 * (128 + 8) + 128 * (32 + 8)) * sizeof(__hip_bfloat16);
    hipLaunchKernelGGL((fmha_fwd_kernel<128>), grid, dim3(256), shmem, stream, q, k, v, o,
                       lse, N, scale);
  }
}

void launch_fmha_bwd_pre(const __hip_bfloat16* dout, const __hip_bfloat16* o, float* D,
                         long rows, int HD, hipStream_t stream) {
  int grid = (int)min(rows, (long)8192);
  hipLaunchKernelGGL(fmha_bwd_pre_kernel, dim3(grid), dim3(256), 0, stream, dout, o, D,
                     rows, HD);
}

void launch_fmha_bwd_dq(const __hip_bfloat16* q, const __hip_bfloat16* k,
                        const __hip_bfloat16* v, const __hip_bfloat16* dout,
                        const float* lse, const float* D, __hip_bfloat16* dq, int BH,
                        int N, int HD, float scale, hipStream_t stream) {
  dim3 grid(BH, (N + 127) / 128);
  if (HD == 64) {
    size_t shmem = (2 * 32 * (64 + 8) + 64 * (32 + 8)) * sizeof(__hip_bfloat16);
    hipLaunchKernelGGL((fmha_bwd_dq_kernel<64>), grid, dim3(256), shmem, stream, q, k, v,
                       dout, lse, D, dq, N, scale);
  } else if (HD == 128) {
    size_t shmem = (2 * 32 * (128 + 8) + 128 * (32 + 8)) * sizeof(__hip_bfloat16);
    hipLaunchKernelGGL((fmha_bwd_dq_kernel<128>), grid, dim3(256), shmem, stream, q, k, v,
                       dout, lse, D, dq, N, scale);
  }
}

void launch_fmha_bwd_dkv(const __hip_bfloat16* q, const __hip_bfloat16* k,
                         const __hip_bfloat16* v, const __hip_bfloat16* dout,
                         const float* lse, const float* D, __hip_bfloat16* dk,
                         __hip_bfloat16* dv, int BH, int N, int HD, float scale,
                         hipStream_t stream) {
  dim3 grid(BH, (N + 127) / 128);
  if (HD == 64) {
    size_t shmem = 2 * 64 * (32 + 8) * sizeof(__hip_bfloat16) + 2 * 32 * sizeof(float);
    hipLaunchKernelGGL((fmha_bwd_dkv_kernel<64>), grid, dim3(256), shmem, stream, q, k, v,
                       dout, lse, D, dk, dv, N, scale);
  } else if (HD == 128) {
    size_t shmem = 2 * 128 * (32 + 8) * sizeof(__hip_bfloat16) + 2 * 32 * sizeof(float);
    hipLaunchKernelGGL((fmha_bwd_dkv_kernel<128>), grid, dim3(256), shmem, stream, q, k, v,
                       dout, lse, D, dk, dv, N, scale);
  }
}
