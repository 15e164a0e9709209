// Fused-layout CSR edge attention (gfx950) — production path.
//
// Differences from edge_attn.hip (which remains as the reference-layout
// kernel used by the op-level parity tests):
//   * q/k/v/skip live in ONE [N,4H] tensor (output of the fused QKVS GEMM)
//     — one GEMM launch and one gradient buffer instead of four;
//   * the per-edge embedding e_ij = W_e [ifc(a0) ‖ rpc(a1)] is NOT
//     materialized: by linearity it equals P_ifc[a0] + P_rpc[a1] where
//     P_ifc = ifc_table @ W_e[:, :H]^T (a tiny [V,H] table computed once per
//     layer) — the kernel gathers from these L2-resident tables instead of
//     streaming an [E,H] operand from HBM (halves forward edge traffic).
//
// Row mapping: LPR lanes per CSR destination row (default 32 — PERT rows
// average degree ~1.3, so packing 2 rows per wave doubles row throughput of
// the latency-bound per-edge loop); reductions are sub-wave shfl_xor trees.
//
// Column mapping: in VEC mode each sub-wave lane owns VPT CONTIGUOUS columns
// (lane*VPT..) so every per-edge row access is vectorized 16B loads (guide
// G13); otherwise the strided lane+j*64 mapping with per-column guards
// (LPR=64 only).  The mapping is an internal detail — every tensor is
// read/written with the same mapping.
//
// Backward: row kernel writes dq + dskip segments of dqkvs and per-edge
// dek/dev scratch; col kernel (CSC) segment-sums dk/dv into dqkvs; de =
// dek + dev feeds the per-vocab scatter (python side) to produce
// dP_ifc/dP_rpc.  Deterministic — no atomics anywhere.

#include "common.h"
#include <cmath>
#include <cstdlib>

#define WAVES_PER_BLOCK 4

typedef __attribute__((ext_vector_type(4))) float f32x4_t;
typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4_t;

// row-slice accessors for the two column mappings; fp32 and bf16 overloads
// (bf16 rows are the qkvs/dqkvs tensors in bf16-resident-activation mode).
template <int VPT, bool VEC>
struct Slice {
  static __device__ __forceinline__ void load(const float* base, int lane,
                                              int h, float (&dst)[VPT]) {
    if constexpr (VEC) {
#pragma unroll
      for (int q = 0; q < VPT; q += 4)
        *reinterpret_cast<f32x4_t*>(&dst[q]) =
            *reinterpret_cast<const f32x4_t*>(&base[lane * VPT + q]);
    } else {
#pragma unroll
      for (int j = 0; j < VPT; ++j) {
        const int c = lane + j * PERTGNN_WAVE;
        dst[j] = (c < h) ? base[c] : 0.f;
      }
    }
  }
  static __device__ __forceinline__ void load(const __bf16* base, int lane,
                                              int h, float (&dst)[VPT]) {
    if constexpr (VEC) {
#pragma unroll
      for (int q = 0; q < VPT; q += 4) {
        const bf16x4_t v =
            *reinterpret_cast<const bf16x4_t*>(&base[lane * VPT + q]);
#pragma unroll
        for (int u = 0; u < 4; ++u) dst[q + u] = (float)v[u];
      }
    } else {
#pragma unroll
      for (int j = 0; j < VPT; ++j) {
        const int c = lane + j * PERTGNN_WAVE;
        dst[j] = (c < h) ? (float)base[c] : 0.f;
      }
    }
  }
  static __device__ __forceinline__ void load_add(const float* a,
                                                  const float* b, int lane,
                                                  int h, float (&dst)[VPT]) {
    if constexpr (VEC) {
#pragma unroll
      for (int q = 0; q < VPT; q += 4) {
        const f32x4_t va = *reinterpret_cast<const f32x4_t*>(&a[lane * VPT + q]);
        const f32x4_t vb = *reinterpret_cast<const f32x4_t*>(&b[lane * VPT + q]);
#pragma unroll
        for (int u = 0; u < 4; ++u) dst[q + u] = va[u] + vb[u];
      }
    } else {
#pragma unroll
      for (int j = 0; j < VPT; ++j) {
        const int c = lane + j * PERTGNN_WAVE;
        dst[j] = (c < h) ? a[c] + b[c] : 0.f;
      }
    }
  }
  static __device__ __forceinline__ void load_add(const __bf16* a,
                                                  const __bf16* b, int lane,
                                                  int h, float (&dst)[VPT]) {
    if constexpr (VEC) {
#pragma unroll
      for (int q = 0; q < VPT; q += 4) {
        const bf16x4_t va =
            *reinterpret_cast<const bf16x4_t*>(&a[lane * VPT + q]);
        const bf16x4_t vb =
            *reinterpret_cast<const bf16x4_t*>(&b[lane * VPT + q]);
#pragma unroll
        for (int u = 0; u < 4; ++u) dst[q + u] = (float)va[u] + (float)vb[u];
      }
    } else {
#pragma unroll
      for (int j = 0; j < VPT; ++j) {
        const int c = lane + j * PERTGNN_WAVE;
        dst[j] = (c < h) ? (float)a[c] + (float)b[c] : 0.f;
      }
    }
  }
  static __device__ __forceinline__ void store(float* base, int lane, int h,
                                               const float (&src)[VPT]) {
    if constexpr (VEC) {
#pragma unroll
      for (int q = 0; q < VPT; q += 4)
        *reinterpret_cast<f32x4_t*>(&base[lane * VPT + q]) =
            *reinterpret_cast<const f32x4_t*>(&src[q]);
    } else {
#pragma unroll
      for (int j = 0; j < VPT; ++j) {
        const int c = lane + j * PERTGNN_WAVE;
        if (c < h) base[c] = src[j];
      }
    }
  }
  static __device__ __forceinline__ void store(__bf16* base, int lane, int h,
                                               const float (&src)[VPT]) {
    if constexpr (VEC) {
#pragma unroll
      for (int q = 0; q < VPT; q += 4) {
        bf16x4_t v;
#pragma unroll
        for (int u = 0; u < 4; ++u) v[u] = (__bf16)src[q + u];
        *reinterpret_cast<bf16x4_t*>(&base[lane * VPT + q]) = v;
      }
    } else {
#pragma unroll
      for (int j = 0; j < VPT; ++j) {
        const int c = lane + j * PERTGNN_WAVE;
        if (c < h) base[c] = (__bf16)src[j];
      }
    }
  }
};

// LPR = lanes per row: CSR rows average degree ~1.3 on PERT graphs, so a
// full 64-lane wave per row is latency-bound with most lanes idle between
// the few row loads.  LPR<64 packs 64/LPR rows into a wave (each sub-group
// owns h/LPR contiguous columns); reductions become sub-wave shfl_xor trees.
// LPR=32 measured best at H=256 (see attn_lpr below).
template <int VPT, bool VEC, typename QT = float, int LPR = PERTGNN_WAVE,
          typename TO = float, typename PT = float>
__global__ void edge_attn_fused_fwd_kernel(
    const QT* __restrict__ qkvs,  // [N, 4h]
    const PT* __restrict__ pifc,  // [Vi, h] (bf16 in act16 mode — halves
    const PT* __restrict__ prpc,  // [Vr, h]  the per-edge ec gathers)
    const long* __restrict__ ea, int astride,
    const int* __restrict__ row_ptr, const int* __restrict__ csr_src,
    TO* __restrict__ out, float* __restrict__ alpha, int n, int h,
    float scale) {
  using S = Slice<VPT, VEC>;
  static_assert(VEC || LPR == PERTGNN_WAVE, "sub-wave rows need VEC layout");
  constexpr int RPW = PERTGNN_WAVE / LPR;
  const int wid = threadIdx.x / PERTGNN_WAVE;
  const int lane0 = threadIdx.x % PERTGNN_WAVE;
  const int sub = lane0 / LPR;
  const int lane = lane0 % LPR;
  const int row = (blockIdx.x * WAVES_PER_BLOCK + wid) * RPW + sub;
  if (row >= n) return;
  const long ld = 4L * h;

  float qr[VPT], acc[VPT];
  S::load(&qkvs[row * ld], lane, h, qr);
#pragma unroll
  for (int j = 0; j < VPT; ++j) acc[j] = 0.f;

  const int beg = row_ptr[row], end = row_ptr[row + 1];
  float m = -INFINITY, s = 0.f;
  for (int p = beg; p < end; ++p) {
    const long src = csr_src[p];
    const long a0 = ea[(long)p * astride];
    const long a1 = ea[(long)p * astride + 1];
    float ec[VPT], ke[VPT], ve[VPT];
    // all three gathers issue together: the ve load must not sit behind the
    // sub-wave reduce + expf chain (one exposed memory latency per edge)
    S::load_add(&pifc[a0 * h], &prpc[a1 * h], lane, h, ec);
    S::load(&qkvs[src * ld + h], lane, h, ke);
    S::load(&qkvs[src * ld + 2 * h], lane, h, ve);
    float part = 0.f;
#pragma unroll
    for (int j = 0; j < VPT; ++j) part += qr[j] * (ke[j] + ec[j]);
    const float logit = subwave_reduce_sum<LPR>(part) * scale;
    if (lane == (p - beg) % LPR) alpha[p] = logit;
    const float m_new = fmaxf(m, logit);
    const float corr = __expf(m - m_new);
    const float pexp = __expf(logit - m_new);
    s = s * corr + pexp;
#pragma unroll
    for (int j = 0; j < VPT; ++j)
      acc[j] = acc[j] * corr + pexp * (ve[j] + ec[j]);
    m = m_new;
  }

  const float inv_s = (s > 0.f) ? 1.f / s : 0.f;
  float sk[VPT], res[VPT];
  S::load(&qkvs[row * ld + 3 * h], lane, h, sk);
#pragma unroll
  for (int j = 0; j < VPT; ++j) res[j] = acc[j] * inv_s + sk[j];
  S::store(&out[(long)row * h], lane, h, res);
  for (int p = beg + lane; p < end; p += LPR)
    alpha[p] = __expf(alpha[p] - m) * inv_s;
}

template <int VPT, bool VEC, typename QT = float,
          int LPR = PERTGNN_WAVE, typename TG = float, typename PT = float>
__global__ void edge_attn_fused_bwd_row_kernel(
    const TG* __restrict__ g, const QT* __restrict__ qkvs,
    const PT* __restrict__ pifc, const PT* __restrict__ prpc,
    const long* __restrict__ ea, int astride, const float* __restrict__ alpha,
    const int* __restrict__ row_ptr, const int* __restrict__ csr_src,
    QT* __restrict__ dqkvs, float* __restrict__ dal, int n, int h,
    float scale) {
  // SINGLE pass over the row's edges (the old two-pass form re-gathered
  // ec+ke per edge after sdot was known).  Softmax backward re-expressed:
  //   dq = scale * (Sum a*da*(k+e)  -  sdot * Sum a*(k+e))
  // accumulates both sums alongside sdot = Sum a*da; the per-edge logit
  // grad dl = scale*a*(da - sdot) is a SCALAR — dek = dl*q_row and
  // dev = a*g_row are rank-1, so the [E,h] dek/dev scratch tensors are
  // gone: the col kernel regenerates them from (dal, alpha) and row
  // gathers of q/g.
  using S = Slice<VPT, VEC>;
  static_assert(VEC || LPR == PERTGNN_WAVE, "sub-wave rows need VEC layout");
  constexpr int RPW = PERTGNN_WAVE / LPR;
  const int wid = threadIdx.x / PERTGNN_WAVE;
  const int lane0 = threadIdx.x % PERTGNN_WAVE;
  const int lane = lane0 % LPR;
  const int row = (blockIdx.x * WAVES_PER_BLOCK + wid) * RPW + lane0 / LPR;
  if (row >= n) return;
  const long ld = 4L * h;

  float gr[VPT], a1[VPT], a2[VPT];
  S::load(&g[(long)row * h], lane, h, gr);
#pragma unroll
  for (int j = 0; j < VPT; ++j) { a1[j] = 0.f; a2[j] = 0.f; }

  const int beg = row_ptr[row], end = row_ptr[row + 1];
  float sdot = 0.f;
  for (int p = beg; p < end; ++p) {
    const long src = csr_src[p];
    const long e0 = ea[(long)p * astride];
    const long e1 = ea[(long)p * astride + 1];
    float ec[VPT], ke[VPT], ve[VPT];
    S::load_add(&pifc[e0 * h], &prpc[e1 * h], lane, h, ec);
    S::load(&qkvs[src * ld + h], lane, h, ke);
    S::load(&qkvs[src * ld + 2 * h], lane, h, ve);
    float part = 0.f;
#pragma unroll
    for (int j = 0; j < VPT; ++j) part += gr[j] * (ve[j] + ec[j]);
    const float dalpha = subwave_reduce_sum<LPR>(part);
    const float a = alpha[p];
    sdot += a * dalpha;
    if (lane == (p - beg) % LPR) dal[p] = dalpha;
#pragma unroll
    for (int j = 0; j < VPT; ++j) {
      const float kec = ke[j] + ec[j];
      a1[j] += a * dalpha * kec;
      a2[j] += a * kec;
    }
  }
  float dq[VPT];
#pragma unroll
  for (int j = 0; j < VPT; ++j) dq[j] = scale * (a1[j] - sdot * a2[j]);
  S::store(&dqkvs[row * ld], lane, h, dq);             // dq
  S::store(&dqkvs[row * ld + 3 * h], lane, h, gr);     // dskip = g
  // finalize per-edge logit grads: each lane rewrites exactly the slots it
  // stored above (same-thread read-after-write)
  for (int p = beg + lane; p < end; p += LPR)
    dal[p] = scale * alpha[p] * (dal[p] - sdot);
}

template <int VPT, bool VEC, typename QT = float, typename ET = float,
          int LPR = PERTGNN_WAVE, typename TG = float>
__global__ void edge_attn_fused_bwd_col_kernel(
    const TG* __restrict__ g, const QT* __restrict__ qkvs,
    const float* __restrict__ alpha, const float* __restrict__ dal,
    const int* __restrict__ col_ptr, const int* __restrict__ csc_dst,
    const int* __restrict__ csc_eid, QT* __restrict__ dqkvs,
    ET* __restrict__ de, int n, int h) {
  using S = Slice<VPT, VEC>;
  static_assert(VEC || LPR == PERTGNN_WAVE, "sub-wave rows need VEC layout");
  constexpr int RPW = PERTGNN_WAVE / LPR;
  const int wid = threadIdx.x / PERTGNN_WAVE;
  const int lane0 = threadIdx.x % PERTGNN_WAVE;
  const int lane = lane0 % LPR;
  const int row = (blockIdx.x * WAVES_PER_BLOCK + wid) * RPW + lane0 / LPR;
  if (row >= n) return;
  const long ld = 4L * h;
  float ka[VPT], va[VPT];
#pragma unroll
  for (int j = 0; j < VPT; ++j) { ka[j] = 0.f; va[j] = 0.f; }
  for (int p = col_ptr[row]; p < col_ptr[row + 1]; ++p) {
    const long eid = csc_eid[p];
    const long dst = csc_dst[p];
    const float dl = dal[eid];
    const float a = alpha[eid];
    float qd[VPT], gd[VPT], des[VPT];
    S::load(&qkvs[dst * ld], lane, h, qd);
    S::load(&g[dst * h], lane, h, gd);
#pragma unroll
    for (int j = 0; j < VPT; ++j) {
      const float dk1 = dl * qd[j];
      const float dv1 = a * gd[j];
      ka[j] += dk1;
      va[j] += dv1;
      des[j] = dk1 + dv1;
    }
    // every edge appears exactly once in the CSC sweep: de rides along
    S::store(&de[eid * h], lane, h, des);
  }
  S::store(&dqkvs[(long)row * ld + h], lane, h, ka);
  S::store(&dqkvs[(long)row * ld + 2 * h], lane, h, va);
}


// ---------------------------------------------------------------------------

void launch_edge_attn_fused_fwd(const float* qkvs, const float* pifc,
                                const float* prpc, const long* ea, int astride,
                                const int* row_ptr, const int* csr_src,
                                float* out, float* alpha, int n, int h,
                                hipStream_t stream) {
  if (n == 0) return;
  const float scale = 1.f / std::sqrt((float)h);
  const dim3 grid(ceil_div(n, WAVES_PER_BLOCK));
  const dim3 block(WAVES_PER_BLOCK * PERTGNN_WAVE);
  const int vpt = (h + PERTGNN_WAVE - 1) / PERTGNN_WAVE;
  const bool vec = (h % 256 == 0);
  switch (vpt) {
#define CASE(V)                                                                \
  case V:                                                                      \
    if (vec && (V % 4 == 0))                                                   \
      edge_attn_fused_fwd_kernel<V, true><<<grid, block, 0, stream>>>(         \
          qkvs, pifc, prpc, ea, astride, row_ptr, csr_src, out, alpha, n, h,   \
          scale);                                                              \
    else                                                                       \
      edge_attn_fused_fwd_kernel<V, false><<<grid, block, 0, stream>>>(        \
          qkvs, pifc, prpc, ea, astride, row_ptr, csr_src, out, alpha, n, h,   \
          scale);                                                              \
    break;
    CASE(1) CASE(2) CASE(3) CASE(4) CASE(5) CASE(6) CASE(7) CASE(8)
#undef CASE
    default: pertgnn_shape_fail("edge_attn_fused launcher", "vpt", vpt);
  }
}

void launch_edge_attn_fused_bwd(const float* g, const float* qkvs,
                                const float* pifc, const float* prpc,
                                const long* ea, int astride,
                                const float* alpha, const int* row_ptr,
                                const int* csr_src, const int* col_ptr,
                                const int* csc_dst, const int* csc_eid,
                                float* dqkvs, float* de, float* dal, int n,
                                int h, long num_edges, hipStream_t stream) {
  if (n == 0) return;
  const float scale = 1.f / std::sqrt((float)h);
  const dim3 grid(ceil_div(n, WAVES_PER_BLOCK));
  const dim3 block(WAVES_PER_BLOCK * PERTGNN_WAVE);
  const int vpt = (h + PERTGNN_WAVE - 1) / PERTGNN_WAVE;
  const bool vec = (h % 256 == 0);
  switch (vpt) {
#define CASE(V)                                                                \
  case V:                                                                      \
    if (vec && (V % 4 == 0)) {                                                 \
      edge_attn_fused_bwd_row_kernel<V, true><<<grid, block, 0, stream>>>(     \
          g, qkvs, pifc, prpc, ea, astride, alpha, row_ptr, csr_src, dqkvs,    \
          dal, n, h, scale);                                                   \
      edge_attn_fused_bwd_col_kernel<V, true><<<grid, block, 0, stream>>>(     \
          g, qkvs, alpha, dal, col_ptr, csc_dst, csc_eid, dqkvs, de, n, h);    \
    } else {                                                                   \
      edge_attn_fused_bwd_row_kernel<V, false><<<grid, block, 0, stream>>>(    \
          g, qkvs, pifc, prpc, ea, astride, alpha, row_ptr, csr_src, dqkvs,    \
          dal, n, h, scale);                                                   \
      edge_attn_fused_bwd_col_kernel<V, false><<<grid, block, 0, stream>>>(    \
          g, qkvs, alpha, dal, col_ptr, csc_dst, csc_eid, dqkvs, de, n, h);    \
    }                                                                          \
    break;
    CASE(1) CASE(2) CASE(3) CASE(4) CASE(5) CASE(6) CASE(7) CASE(8)
#undef CASE
    default: pertgnn_shape_fail("edge_attn_fused launcher", "vpt", vpt);
  }
}


// ---------------------------------------------------------------------------
// bf16-qkvs launchers (activation-resident bf16 mode; requires h % 256 == 0)
// ---------------------------------------------------------------------------

// sub-wave row packing: PERT CSR rows average degree ~1.3, so a full wave
// per row is latency-bound with idle lanes.  LPR=32 (two rows per wave,
// 8 cols/lane at H=256) measured best: 14.87 vs 15.33 ms/step at LPR=64;
// LPR=16 loses it back to register pressure.  PERTGNN_ATTN_LPR overrides.
static int attn_lpr() {
  static int v = [] {
    const char* e = getenv("PERTGNN_ATTN_LPR");
    const int x = e ? atoi(e) : 32;
    return (x == 16 || x == 32 || x == 64) ? x : 32;
  }();
  return v;
}

void launch_edge_attn_fused_fwd16(const void* qkvs_v, const void* pifc_v,
                                  const void* prpc_v, int p16, const long* ea,
                                  int astride, const int* row_ptr,
                                  const int* csr_src, void* out_v, int out16,
                                  float* alpha, int n, int h,
                                  hipStream_t stream) {
  const __bf16* qkvs = (const __bf16*)qkvs_v;
  if (n == 0) return;
  const float scale = 1.f / std::sqrt((float)h);
  const dim3 block(WAVES_PER_BLOCK * PERTGNN_WAVE);
  const int lpr = attn_lpr();
  const int rpb = WAVES_PER_BLOCK * (PERTGNN_WAVE / lpr);
  const dim3 grid(ceil_div(n, rpb));
#define FWD16(VPT, LPR)                                                        \
  do {                                                                         \
    if (p16 && out16) {                                                        \
      edge_attn_fused_fwd_kernel<VPT, true, __bf16, LPR, __bf16, __bf16>       \
          <<<grid, block, 0, stream>>>(qkvs, (const __bf16*)pifc_v,            \
                                       (const __bf16*)prpc_v, ea, astride,     \
                                       row_ptr, csr_src, (__bf16*)out_v,       \
                                       alpha, n, h, scale);                    \
    } else if (p16) {                                                          \
      edge_attn_fused_fwd_kernel<VPT, true, __bf16, LPR, float, __bf16>        \
          <<<grid, block, 0, stream>>>(qkvs, (const __bf16*)pifc_v,            \
                                       (const __bf16*)prpc_v, ea, astride,     \
                                       row_ptr, csr_src, (float*)out_v,        \
                                       alpha, n, h, scale);                    \
    } else if (out16) {                                                        \
      edge_attn_fused_fwd_kernel<VPT, true, __bf16, LPR, __bf16>               \
          <<<grid, block, 0, stream>>>(qkvs, (const float*)pifc_v,             \
                                       (const float*)prpc_v, ea, astride,      \
                                       row_ptr, csr_src, (__bf16*)out_v,       \
                                       alpha, n, h, scale);                    \
    } else {                                                                   \
      edge_attn_fused_fwd_kernel<VPT, true, __bf16, LPR, float>                \
          <<<grid, block, 0, stream>>>(qkvs, (const float*)pifc_v,             \
                                       (const float*)prpc_v, ea, astride,      \
                                       row_ptr, csr_src, (float*)out_v,        \
                                       alpha, n, h, scale);                    \
    }                                                                          \
  } while (0)
  if (h == 256) {
    if (lpr == 16) FWD16(16, 16);
    else if (lpr == 32) FWD16(8, 32);
    else FWD16(4, 64);
  } else if (h == 512) {
    if (lpr == 16 || lpr == 32) FWD16(16, 32);
    else FWD16(8, 64);
  } else if (h % 256 == 0 && h / PERTGNN_WAVE <= 32) {
    const int vpt = h / PERTGNN_WAVE;
    if (vpt == 4) FWD16(4, 64);
    else if (vpt == 8) FWD16(8, 64);
    else if (vpt == 16) FWD16(16, 64);
    else FWD16(32, 64);
  } else {
    pertgnn_shape_fail("edge_attn_fused act16 launcher", "h", h);
  }
#undef FWD16
}

void launch_edge_attn_fused_bwd16(const void* g_v, int g16,
                                  const void* qkvs_v,
                                  const void* pifc_v, const void* prpc_v,
                                  int p16, const long* ea, int astride,
                                  const float* alpha, const int* row_ptr,
                                  const int* csr_src, const int* col_ptr,
                                  const int* csc_dst, const int* csc_eid,
                                  void* dqkvs_v, void* de_v, float* dal,
                                  int n, int h, long num_edges,
                                  hipStream_t stream) {
  const __bf16* qkvs = (const __bf16*)qkvs_v;
  __bf16* dqkvs = (__bf16*)dqkvs_v;
  __bf16* de = (__bf16*)de_v;
  if (n == 0) return;
  const float scale = 1.f / std::sqrt((float)h);
  const dim3 block(WAVES_PER_BLOCK * PERTGNN_WAVE);
  const int lpr = attn_lpr();
  const int rpb = WAVES_PER_BLOCK * (PERTGNN_WAVE / lpr);
  const dim3 grid(ceil_div(n, rpb));
#define BWD16(VPT, LPR)                                                        \
  do {                                                                         \
    if (g16 && p16) {                                                          \
      edge_attn_fused_bwd_row_kernel<VPT, true, __bf16, LPR, __bf16, __bf16>   \
          <<<grid, block, 0, stream>>>((const __bf16*)g_v, qkvs,               \
                                       (const __bf16*)pifc_v,                  \
                                       (const __bf16*)prpc_v,                  \
                                       ea, astride, alpha, row_ptr, csr_src,   \
                                       dqkvs, dal, n, h, scale);               \
      edge_attn_fused_bwd_col_kernel<VPT, true, __bf16, __bf16, LPR, __bf16>   \
          <<<grid, block, 0, stream>>>((const __bf16*)g_v, qkvs, alpha, dal,   \
                                       col_ptr, csc_dst, csc_eid, dqkvs, de,   \
                                       n, h);                                  \
    } else if (g16) {                                                          \
      edge_attn_fused_bwd_row_kernel<VPT, true, __bf16, LPR, __bf16>           \
          <<<grid, block, 0, stream>>>((const __bf16*)g_v, qkvs,               \
                                       (const float*)pifc_v,                   \
                                       (const float*)prpc_v,                   \
                                       ea, astride, alpha, row_ptr, csr_src,   \
                                       dqkvs, dal, n, h, scale);               \
      edge_attn_fused_bwd_col_kernel<VPT, true, __bf16, __bf16, LPR, __bf16>   \
          <<<grid, block, 0, stream>>>((const __bf16*)g_v, qkvs, alpha, dal,   \
                                       col_ptr, csc_dst, csc_eid, dqkvs, de,   \
                                       n, h);                                  \
    } else if (p16) {                                                          \
      /* fp32 upstream grad (the last conv's out16=false) + bf16 P tables — \
         reading the tables at the wrong width over-reads 2x (OOB) */        \
      edge_attn_fused_bwd_row_kernel<VPT, true, __bf16, LPR, float, __bf16>    \
          <<<grid, block, 0, stream>>>((const float*)g_v, qkvs,                \
                                       (const __bf16*)pifc_v,                  \
                                       (const __bf16*)prpc_v,                  \
                                       ea, astride, alpha, row_ptr, csr_src,   \
                                       dqkvs, dal, n, h, scale);               \
      edge_attn_fused_bwd_col_kernel<VPT, true, __bf16, __bf16, LPR, float>    \
          <<<grid, block, 0, stream>>>((const float*)g_v, qkvs, alpha, dal,    \
                                       col_ptr, csc_dst, csc_eid, dqkvs, de,   \
                                       n, h);                                  \
    } else {                                                                   \
      edge_attn_fused_bwd_row_kernel<VPT, true, __bf16, LPR, float>            \
          <<<grid, block, 0, stream>>>((const float*)g_v, qkvs,                \
                                       (const float*)pifc_v,                   \
                                       (const float*)prpc_v,                   \
                                       ea, astride, alpha, row_ptr, csr_src,   \
                                       dqkvs, dal, n, h, scale);               \
      edge_attn_fused_bwd_col_kernel<VPT, true, __bf16, __bf16, LPR, float>    \
          <<<grid, block, 0, stream>>>((const float*)g_v, qkvs, alpha, dal,    \
                                       col_ptr, csc_dst, csc_eid, dqkvs, de,   \
                                       n, h);                                  \
    }                                                                          \
  } while (0)
  if (h == 256) {
    if (lpr == 16) BWD16(16, 16);
    else if (lpr == 32) BWD16(8, 32);
    else BWD16(4, 64);
  } else if (h == 512) {
    if (lpr == 16 || lpr == 32) BWD16(16, 32);
    else BWD16(8, 64);
  } else if (h % 256 == 0 && h / PERTGNN_WAVE <= 32) {
    const int vpt = h / PERTGNN_WAVE;
    if (vpt == 4) BWD16(4, 64);
    else if (vpt == 8) BWD16(8, 64);
    else if (vpt == 16) BWD16(16, 64);
    else BWD16(32, 64);
  } else {
    pertgnn_shape_fail("edge_attn_fused act16 launcher", "h", h);
  }
#undef BWD16
}
