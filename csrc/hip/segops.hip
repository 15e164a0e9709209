// Segment / elementwise / reduction kernels (gfx950):
//   K9+K10  pattern-weighted global-add-pool (fwd: wave-per-graph segment sum
//           over the batch-ptr — deterministic, no atomics; bwd: gather)
//   K1+K11  embedding gathers fused with concat (+ scatter-add backward)
//   K7+K8   BatchNorm1d (+fused ReLU) forward/backward
//   K12/K13 quantile loss + eval-metric reductions
//   K14     Adam step (flat master buffers — one kernel for the whole model)
#include "common.h"
#include <cstdlib>

// ---------------------------------------------------------------------------
// pattern pool: out[b] = sum_{i in graph b} x[i] * p[i] / n[i]
// ---------------------------------------------------------------------------

#define WAVES_PER_BLOCK 4

// two-level: P sub-waves per graph each sum a strided node interleave into
// partial[g*P+sub], then one wave per graph folds the P partials in order
// (deterministic, no atomics; single waves per graph would leave the chip
// ~64/256 CUs busy at trace-scale batches).
template <int VPT, typename TX = float>
__global__ void seg_pool_p1_kernel(const TX* __restrict__ x,
                                   const float* __restrict__ probs,
                                   const float* __restrict__ nn,
                                   const int* __restrict__ batch_ptr,
                                   float* __restrict__ partial, int b, int P,
                                   int h) {
  const int wid = threadIdx.x / PERTGNN_WAVE;
  const int lane = threadIdx.x % PERTGNN_WAVE;
  const int w = blockIdx.x * WAVES_PER_BLOCK + wid;
  if (w >= b * P) return;
  const int g = w / P;
  const int sub = w % P;
  float acc[VPT];
#pragma unroll
  for (int j = 0; j < VPT; ++j) acc[j] = 0.f;
  for (int i = batch_ptr[g] + sub; i < batch_ptr[g + 1]; i += P) {
    const float wt = probs[i] / nn[i];
#pragma unroll
    for (int j = 0; j < VPT; ++j) {
      const int c = lane + j * PERTGNN_WAVE;
      if (c < h) acc[j] += (float)x[(long)i * h + c] * wt;
    }
  }
#pragma unroll
  for (int j = 0; j < VPT; ++j) {
    const int c = lane + j * PERTGNN_WAVE;
    if (c < h) partial[(long)w * h + c] = acc[j];
  }
}

template <int VPT>
__global__ void seg_pool_p2_kernel(const float* __restrict__ partial,
                                   float* __restrict__ out, int b, int P,
                                   int h) {
  const int wid = threadIdx.x / PERTGNN_WAVE;
  const int lane = threadIdx.x % PERTGNN_WAVE;
  const int g = blockIdx.x * WAVES_PER_BLOCK + wid;
  if (g >= b) return;
  float acc[VPT];
#pragma unroll
  for (int j = 0; j < VPT; ++j) acc[j] = 0.f;
  for (int sub = 0; sub < P; ++sub) {
#pragma unroll
    for (int j = 0; j < VPT; ++j) {
      const int c = lane + j * PERTGNN_WAVE;
      if (c < h) acc[j] += partial[((long)g * P + sub) * h + c];
    }
  }
#pragma unroll
  for (int j = 0; j < VPT; ++j) {
    const int c = lane + j * PERTGNN_WAVE;
    if (c < h) out[(long)g * h + c] = acc[j];
  }
}

template <typename TD = float>
__global__ void seg_pool_bwd_kernel(const float* __restrict__ gout,
                                    const float* __restrict__ probs,
                                    const float* __restrict__ nn,
                                    const long* __restrict__ batch,
                                    TD* __restrict__ dx, long n, int h) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  const long numel = n * h;
  for (long t = i; t < numel; t += stride) {
    const long row = t / h;
    const int c = (int)(t - row * h);
    dx[t] = (TD)(gout[batch[row] * h + c] * probs[row] / nn[row]);
  }
}

void launch_seg_pool_fwd(const float* x, const float* probs, const float* nn,
                         const int* batch_ptr, float* partial, float* out,
                         int b, int P, int h, hipStream_t stream) {
  if (b == 0) return;
  const dim3 block(WAVES_PER_BLOCK * PERTGNN_WAVE);
  const int vpt = (h + PERTGNN_WAVE - 1) / PERTGNN_WAVE;
  switch (vpt) {
#define CASE(V)                                                                \
  case V:                                                                      \
    seg_pool_p1_kernel<V>                                                      \
        <<<dim3(ceil_div((long)b * P, WAVES_PER_BLOCK)), block, 0, stream>>>(  \
            x, probs, nn, batch_ptr, partial, b, P, h);                        \
    seg_pool_p2_kernel<V>                                                      \
        <<<dim3(ceil_div(b, WAVES_PER_BLOCK)), block, 0, stream>>>(            \
            partial, out, b, P, h);                                            \
    break;
    CASE(1) CASE(2) CASE(3) CASE(4) CASE(5) CASE(6) CASE(7) CASE(8)
#undef CASE
    default: pertgnn_shape_fail("segops launcher", "vpt", vpt);
  }
}

void launch_seg_pool_bwd(const float* gout, const float* probs,
                         const float* nn, const long* batch, float* dx, long n,
                         int h, hipStream_t stream) {
  if (n == 0) return;
  const long numel = n * h;
  const int tpb = 256;
  const int blocks = (int)min((numel + tpb - 1) / tpb, (long)4096);
  seg_pool_bwd_kernel<<<dim3(blocks), dim3(tpb), 0, stream>>>(gout, probs, nn,
                                                              batch, dx, n, h);
}

// act16 variants: x16 in (bf16 last-layer activations), bf16 dx out
void launch_seg_pool_fwd16(const void* x, const float* probs, const float* nn,
                           const int* batch_ptr, float* partial, float* out,
                           int b, int P, int h, hipStream_t stream) {
  if (b == 0) return;
  const dim3 block(WAVES_PER_BLOCK * PERTGNN_WAVE);
  const int vpt = (h + PERTGNN_WAVE - 1) / PERTGNN_WAVE;
  switch (vpt) {
#define CASE(V)                                                                \
  case V:                                                                      \
    seg_pool_p1_kernel<V, __bf16>                                              \
        <<<dim3(ceil_div((long)b * P, WAVES_PER_BLOCK)), block, 0, stream>>>(  \
            (const __bf16*)x, probs, nn, batch_ptr, partial, b, P, h);         \
    seg_pool_p2_kernel<V>                                                      \
        <<<dim3(ceil_div(b, WAVES_PER_BLOCK)), block, 0, stream>>>(            \
            partial, out, b, P, h);                                            \
    break;
    CASE(1) CASE(2) CASE(3) CASE(4) CASE(5) CASE(6) CASE(7) CASE(8)
#undef CASE
    default: pertgnn_shape_fail("segops launcher", "vpt", vpt);
  }
}

void launch_seg_pool_bwd16(const float* gout, const float* probs,
                           const float* nn, const long* batch, void* dx,
                           long n, int h, hipStream_t stream) {
  if (n == 0) return;
  const long numel = n * h;
  const int tpb = 256;
  const int blocks = (int)min((numel + tpb - 1) / tpb, (long)4096);
  seg_pool_bwd_kernel<<<dim3(blocks), dim3(tpb), 0, stream>>>(
      gout, probs, nn, batch, (__bf16*)dx, n, h);
}

// ---------------------------------------------------------------------------
// embedding gathers fused with concat
// ---------------------------------------------------------------------------

template <typename TY = float>
__global__ void embed_node_fwd_kernel(const float* __restrict__ x_raw,
                                      const long* __restrict__ idx,
                                      const float* __restrict__ table,
                                      TY* __restrict__ out, long n, int f,
                                      int h) {
  const int w = f + h;
  const long numel = n * w;
  const long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long t = i0; t < numel; t += stride) {
    const long row = t / w;
    const int c = (int)(t - row * w);
    out[t] = (TY)((c < f) ? x_raw[row * f + c]
                          : table[idx[row] * h + (c - f)]);
  }
}

__global__ void embed_edge_fwd_kernel(const long* __restrict__ attr,
                                      const float* __restrict__ ifc,
                                      const float* __restrict__ rpc,
                                      float* __restrict__ out, long e, int h,
                                      int astride) {
  const int w = 2 * h;
  const long numel = e * w;
  const long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long t = i0; t < numel; t += stride) {
    const long row = t / w;
    const int c = (int)(t - row * w);
    out[t] = (c < h) ? ifc[attr[row * astride] * h + c]
                     : rpc[attr[row * astride + 1] * h + (c - h)];
  }
}

// Deterministic grouped embedding backward: rows of g pre-grouped by table
// index (order/ptr built once per backward from a sort).  Low-cardinality
// tables (rpctype has ~5 rows over 5e4 edges) would serialize a whole segment
// on one wave, so the reduction is TWO-PHASE: P sub-waves per table row each
// sum a strided interleave of the segment (phase 1, deterministic), then one
// wave per row folds the P partials in fixed order (phase 2).  No atomics.
template <int VPT>
__global__ void embed_grouped_p1_kernel(
    const float* __restrict__ g, const int* __restrict__ order,
    const int* __restrict__ ptr, float* __restrict__ partial, int rows, int P,
    int h, int gstride, int col_off) {
  const int wid = threadIdx.x / PERTGNN_WAVE;
  const int lane = threadIdx.x % PERTGNN_WAVE;
  const int w = blockIdx.x * WAVES_PER_BLOCK + wid;
  if (w >= rows * P) return;
  const int row = w / P;
  const int sub = w % P;
  float acc[VPT];
#pragma unroll
  for (int j = 0; j < VPT; ++j) acc[j] = 0.f;
  for (int p = ptr[row] + sub; p < ptr[row + 1]; p += P) {
    const long r = order[p];
#pragma unroll
    for (int j = 0; j < VPT; ++j) {
      const int c = lane + j * PERTGNN_WAVE;
      if (c < h) acc[j] += g[r * gstride + col_off + c];
    }
  }
#pragma unroll
  for (int j = 0; j < VPT; ++j) {
    const int c = lane + j * PERTGNN_WAVE;
    if (c < h) partial[(long)w * h + c] = acc[j];
  }
}

template <int VPT>
__global__ void embed_grouped_p2_kernel(const float* __restrict__ partial,
                                        float* __restrict__ dtable, int rows,
                                        int P, int h) {
  const int wid = threadIdx.x / PERTGNN_WAVE;
  const int lane = threadIdx.x % PERTGNN_WAVE;
  const int row = blockIdx.x * WAVES_PER_BLOCK + wid;
  if (row >= rows) return;
  float acc[VPT];
#pragma unroll
  for (int j = 0; j < VPT; ++j) acc[j] = 0.f;
  for (int sub = 0; sub < P; ++sub) {
#pragma unroll
    for (int j = 0; j < VPT; ++j) {
      const int c = lane + j * PERTGNN_WAVE;
      if (c < h) acc[j] += partial[((long)row * P + sub) * h + c];
    }
  }
#pragma unroll
  for (int j = 0; j < VPT; ++j) {
    const int c = lane + j * PERTGNN_WAVE;
    if (c < h) dtable[(long)row * h + c] = acc[j];
  }
}

// order==nullptr reads g rows directly (r = p): that is how the SECOND
// reduction level folds the level-1 partials — same balanced assignment,
// with ptr = level-1 wave_start (each row's partials are contiguous).
// VEC: contiguous lane*VPT column mapping with packed 8-B (bf16) / 16-B
// (fp32) row loads — the strided lane+j*64 mapping issues VPT scalar loads
// per gathered row, which dominates this latency-bound kernel.  Requires
// h == VPT*64, VPT%4==0, and 4-element-aligned rows (gstride%4, col_off%4).
// The column mapping is internal: p1, the level-2 fold and p2 all use the
// same c_of(j), so dtable comes out identical either way.
typedef __attribute__((ext_vector_type(4))) float segf4;
typedef __attribute__((ext_vector_type(4))) __bf16 segb4;

template <int VPT, typename TG = float, bool VEC = false>
__global__ void embed_grouped_bal_p1_kernel(
    const TG* __restrict__ g, const int* __restrict__ order,
    const int* __restrict__ ptr, const int* __restrict__ row_map,
    const int* __restrict__ wave_start, float* __restrict__ partial,
    int n_waves, int h, int gstride, int col_off) {
  const int wid = threadIdx.x / PERTGNN_WAVE;
  const int lane = threadIdx.x % PERTGNN_WAVE;
  const int w = blockIdx.x * WAVES_PER_BLOCK + wid;
  if (w >= n_waves) return;
  const int row = row_map[w];
  const int w0 = wave_start[row];
  const int P = wave_start[row + 1] - w0;
  const int sub = w - w0;
  float acc[VPT];
#pragma unroll
  for (int j = 0; j < VPT; ++j) acc[j] = 0.f;
  for (int p = ptr[row] + sub; p < ptr[row + 1]; p += P) {
    const long r = order ? order[p] : p;
    const TG* src = &g[(long)r * gstride + col_off];
    if constexpr (VEC) {
#pragma unroll
      for (int q = 0; q < VPT; q += 4) {
        if constexpr (sizeof(TG) == 2) {
          const segb4 v = *reinterpret_cast<const segb4*>(
              (const __bf16*)src + lane * VPT + q);
#pragma unroll
          for (int u = 0; u < 4; ++u) acc[q + u] += (float)v[u];
        } else {
          const segf4 v = *reinterpret_cast<const segf4*>(
              (const float*)src + lane * VPT + q);
#pragma unroll
          for (int u = 0; u < 4; ++u) acc[q + u] += v[u];
        }
      }
    } else {
#pragma unroll
      for (int j = 0; j < VPT; ++j) {
        const int c = lane + j * PERTGNN_WAVE;
        if (c < h) acc[j] += (float)src[c];
      }
    }
  }
#pragma unroll
  for (int j = 0; j < VPT; ++j) {
    const int c = VEC ? lane * VPT + j : lane + j * PERTGNN_WAVE;
    if (c < h) partial[(long)w * h + c] = acc[j];
  }
}

template <int VPT, bool VEC = false>
__global__ void embed_grouped_bal_p2_kernel(
    const float* __restrict__ partial, const int* __restrict__ wave_start,
    float* __restrict__ dtable, int rows, int h) {
  const int wid = threadIdx.x / PERTGNN_WAVE;
  const int lane = threadIdx.x % PERTGNN_WAVE;
  const int row = blockIdx.x * WAVES_PER_BLOCK + wid;
  if (row >= rows) return;
  float acc[VPT];
#pragma unroll
  for (int j = 0; j < VPT; ++j) acc[j] = 0.f;
  for (int s = wave_start[row]; s < wave_start[row + 1]; ++s) {
    if constexpr (VEC) {
#pragma unroll
      for (int q = 0; q < VPT; q += 4) {
        const segf4 v = *reinterpret_cast<const segf4*>(
            &partial[(long)s * h + lane * VPT + q]);
#pragma unroll
        for (int u = 0; u < 4; ++u) acc[q + u] += v[u];
      }
    } else {
#pragma unroll
      for (int j = 0; j < VPT; ++j) {
        const int c = lane + j * PERTGNN_WAVE;
        if (c < h) acc[j] += partial[(long)s * h + c];
      }
    }
  }
#pragma unroll
  for (int j = 0; j < VPT; ++j) {
    const int c = VEC ? lane * VPT + j : lane + j * PERTGNN_WAVE;
    if (c < h) dtable[(long)row * h + c] = acc[j];
  }
}

// Two-level balanced reduction: level 1 over the grouped g rows, optional
// level 2 over the level-1 partials (needed when one group's partial count
// itself is large — the interface-0 mega-group), final fold per row.
void launch_embed_grouped_scatter_bal(
    const void* g, int g16, const int* order, const int* ptr,
    const int* row_map, const int* wave_start, const int* row_map2,
    const int* wave_start2, float* partial, float* partial2, float* dtable,
    int n_waves, int n_waves2, int rows, int h, int gstride, int col_off,
    hipStream_t s) {
  if (rows == 0) return;
  const dim3 block(WAVES_PER_BLOCK * PERTGNN_WAVE);
  const int vpt = (h + PERTGNN_WAVE - 1) / PERTGNN_WAVE;
  // packed-load column mapping when the whole row divides evenly and the
  // g rows are 4-element aligned (the hot de/entry cases; the embed-node
  // grad with col_off=9 keeps the strided scalar path)
  const bool vec = (h == vpt * PERTGNN_WAVE) && (vpt % 4 == 0) &&
                   (gstride % 4 == 0) && (col_off % 4 == 0);
  switch (vpt) {
#define CASE(V) case V: { const int grid1 = ceil_div(n_waves, WAVES_PER_BLOCK); if (vec) { if (g16) { embed_grouped_bal_p1_kernel<V, __bf16, true><<<dim3(grid1), block, 0, s>>>((const __bf16*)g, order, ptr, row_map, wave_start, partial, n_waves, h, gstride, col_off); } else { embed_grouped_bal_p1_kernel<V, float, true><<<dim3(grid1), block, 0, s>>>((const float*)g, order, ptr, row_map, wave_start, partial, n_waves, h, gstride, col_off); } if (n_waves2 > 0) { embed_grouped_bal_p1_kernel<V, float, true><<<dim3(ceil_div(n_waves2, WAVES_PER_BLOCK)), block, 0, s>>>(partial, nullptr, wave_start, row_map2, wave_start2, partial2, n_waves2, h, h, 0); embed_grouped_bal_p2_kernel<V, true><<<dim3(ceil_div(rows, WAVES_PER_BLOCK)), block, 0, s>>>(partial2, wave_start2, dtable, rows, h); } else { embed_grouped_bal_p2_kernel<V, true><<<dim3(ceil_div(rows, WAVES_PER_BLOCK)), block, 0, s>>>(partial, wave_start, dtable, rows, h); } } else { if (g16) { embed_grouped_bal_p1_kernel<V, __bf16><<<dim3(grid1), block, 0, s>>>((const __bf16*)g, order, ptr, row_map, wave_start, partial, n_waves, h, gstride, col_off); } else { embed_grouped_bal_p1_kernel<V, float><<<dim3(grid1), block, 0, s>>>((const float*)g, order, ptr, row_map, wave_start, partial, n_waves, h, gstride, col_off); } if (n_waves2 > 0) { embed_grouped_bal_p1_kernel<V, float><<<dim3(ceil_div(n_waves2, WAVES_PER_BLOCK)), block, 0, s>>>(partial, nullptr, wave_start, row_map2, wave_start2, partial2, n_waves2, h, h, 0); embed_grouped_bal_p2_kernel<V><<<dim3(ceil_div(rows, WAVES_PER_BLOCK)), block, 0, s>>>(partial2, wave_start2, dtable, rows, h); } else { embed_grouped_bal_p2_kernel<V><<<dim3(ceil_div(rows, WAVES_PER_BLOCK)), block, 0, s>>>(partial, wave_start, dtable, rows, h); } } } break;
    CASE(1) CASE(2) CASE(3) CASE(4) CASE(5) CASE(6) CASE(7) CASE(8)
#undef CASE
    default: pertgnn_shape_fail("segops launcher", "vpt", vpt);
  }
}

void launch_embed_grouped_scatter(const float* g, const int* order,
                                  const int* ptr, float* partial,
                                  float* dtable, long num_src, int rows, int P,
                                  int h, int gstride, int col_off,
                                  hipStream_t s) {
  if (rows == 0) return;
  const dim3 block(WAVES_PER_BLOCK * PERTGNN_WAVE);
  const int vpt = (h + PERTGNN_WAVE - 1) / PERTGNN_WAVE;
  switch (vpt) {
#define CASE(V)                                                                \
  case V:                                                                      \
    embed_grouped_p1_kernel<V>                                                 \
        <<<dim3(ceil_div((long)rows * P, WAVES_PER_BLOCK)), block, 0, s>>>(    \
            g, order, ptr, partial, rows, P, h, gstride, col_off);             \
    embed_grouped_p2_kernel<V>                                                 \
        <<<dim3(ceil_div(rows, WAVES_PER_BLOCK)), block, 0, s>>>(              \
            partial, dtable, rows, P, h);                                      \
    break;
    CASE(1) CASE(2) CASE(3) CASE(4) CASE(5) CASE(6) CASE(7) CASE(8)
#undef CASE
    default: pertgnn_shape_fail("segops launcher", "vpt", vpt);
  }
}

// Single-pass vocab accumulator for small vocabularies: blocks stream
// CONTIGUOUS gradient rows (coalesced, no sort/gather), accumulate into an
// LDS-resident [V,h] table (LDS atomics — contention only within the CU),
// then one global atomicAdd per (v,c) per block.  Weight-gradient class:
// fp32 reduction order varies run to run (same class as split-K wgrad).
__global__ void vocab_scatter_kernel(const float* __restrict__ g,
                                     const long* __restrict__ idx,
                                     long idx_stride, float* __restrict__ dtable,
                                     long n, int rows, int h, int gstride,
                                     int col_off) {
  extern __shared__ float acc[];  // [rows*h]
  const long vh = (long)rows * h;
  for (long t = threadIdx.x; t < vh; t += blockDim.x) acc[t] = 0.f;
  __syncthreads();
  const int wid = threadIdx.x / PERTGNN_WAVE;
  const int lane = threadIdx.x % PERTGNN_WAVE;
  const long rows_per_block = (n + gridDim.x - 1) / gridDim.x;
  const long r0 = (long)blockIdx.x * rows_per_block;
  const long r1 = min(n, r0 + rows_per_block);
  for (long r = r0 + wid; r < r1; r += WAVES_PER_BLOCK) {
    const long v = idx[r * idx_stride];
    for (int c = lane; c < h; c += PERTGNN_WAVE)
      atomicAdd(&acc[v * h + c], g[r * gstride + col_off + c]);
  }
  __syncthreads();
  for (long t = threadIdx.x; t < vh; t += blockDim.x)
    if (acc[t] != 0.f) atomicAdd(&dtable[t], acc[t]);
}

// vectorized variant (h % 256 == 0, aligned): lane owns 4 contiguous columns
// — one f32x4 load per row instead of 4 strided scalars (measured: loads, not
// the LDS atomics, dominate this kernel).
__global__ void vocab_scatter_vec_kernel(const float* __restrict__ g,
                                         const long* __restrict__ idx,
                                         long idx_stride,
                                         float* __restrict__ dtable, long n,
                                         int rows, int h, int gstride,
                                         int col_off) {
  typedef __attribute__((ext_vector_type(4))) float f4;
  extern __shared__ float acc[];  // [rows*h]
  const long vh = (long)rows * h;
  for (long t = threadIdx.x; t < vh; t += blockDim.x) acc[t] = 0.f;
  __syncthreads();
  const int wid = threadIdx.x / PERTGNN_WAVE;
  const int lane = threadIdx.x % PERTGNN_WAVE;
  const long rows_per_block = (n + gridDim.x - 1) / gridDim.x;
  const long r0 = (long)blockIdx.x * rows_per_block;
  const long r1 = min(n, r0 + rows_per_block);
  const int nq = h / (4 * PERTGNN_WAVE);  // f32x4 chunks per lane (1 at h=256)
  long r = r0 + wid;
  for (; r + WAVES_PER_BLOCK <= r1; r += 2 * WAVES_PER_BLOCK) {
    const long ra = r, rb = r + WAVES_PER_BLOCK;
    const long va = idx[ra * idx_stride];
    const long vb = idx[rb * idx_stride];
    for (int q = 0; q < nq; ++q) {
      const int c = (q * PERTGNN_WAVE + lane) * 4;
      const f4 xa = *reinterpret_cast<const f4*>(&g[ra * gstride + col_off + c]);
      const f4 xb = *reinterpret_cast<const f4*>(&g[rb * gstride + col_off + c]);
#pragma unroll
      for (int u = 0; u < 4; ++u) atomicAdd(&acc[va * h + c + u], xa[u]);
#pragma unroll
      for (int u = 0; u < 4; ++u) atomicAdd(&acc[vb * h + c + u], xb[u]);
    }
  }
  for (; r < r1; r += WAVES_PER_BLOCK) {
    const long v = idx[r * idx_stride];
    for (int q = 0; q < nq; ++q) {
      const int c = (q * PERTGNN_WAVE + lane) * 4;
      const f4 xv = *reinterpret_cast<const f4*>(&g[r * gstride + col_off + c]);
#pragma unroll
      for (int u = 0; u < 4; ++u) atomicAdd(&acc[v * h + c + u], xv[u]);
    }
  }
  __syncthreads();
  for (long t = threadIdx.x; t < vh; t += blockDim.x)
    if (acc[t] != 0.f) atomicAdd(&dtable[t], acc[t]);
}

// dual-table variant: accumulates BOTH vocab tables (interface + rpctype)
// in one pass over g — the two dP reductions of the attention backward share
// the same 56MB de stream.
typedef __attribute__((ext_vector_type(4))) float vs_f4;
typedef __attribute__((ext_vector_type(4))) __bf16 vs_b4;
__device__ __forceinline__ void vs_ld4(const float* p, float (&v)[4]) {
  *reinterpret_cast<vs_f4*>(v) = *reinterpret_cast<const vs_f4*>(p);
}
__device__ __forceinline__ void vs_ld4(const __bf16* p, float (&v)[4]) {
  const vs_b4 b = *reinterpret_cast<const vs_b4*>(p);
#pragma unroll
  for (int u = 0; u < 4; ++u) v[u] = (float)b[u];
}

// U rows are prefetched per loop iteration (independent index + gradient
// loads issue together before the LDS-atomic block) — the plain one-row loop
// exposes two dependent memory latencies per row and, at 1 block/CU, ran at
// ~200 GB/s; prefetch + 3 blocks/CU recovers the latency-bound gap.
template <typename GT, int U>
__global__ void vocab_scatter_dual_kernel(const GT* __restrict__ g,
                                          const long* __restrict__ ea,
                                          int astride,
                                          float* __restrict__ dt0,
                                          float* __restrict__ dt1, long n,
                                          int rows0, int rows1, int h) {
  extern __shared__ float acc[];  // [(rows0+rows1)*h]
  float* acc1 = acc + (long)rows0 * h;
  const long vh = (long)(rows0 + rows1) * h;
  for (long t = threadIdx.x; t < vh; t += blockDim.x) acc[t] = 0.f;
  __syncthreads();
  const int wid = threadIdx.x / PERTGNN_WAVE;
  const int lane = threadIdx.x % PERTGNN_WAVE;
  const long rows_per_block = (n + gridDim.x - 1) / gridDim.x;
  const long r0 = (long)blockIdx.x * rows_per_block;
  const long r1 = min(n, r0 + rows_per_block);
  const int nq = h / (4 * PERTGNN_WAVE);
  const long step = WAVES_PER_BLOCK;
  long r = r0 + wid;
  for (; r + (U - 1) * step < r1; r += U * step) {
    long v0[U], v1[U];
#pragma unroll
    for (int u = 0; u < U; ++u) {
      v0[u] = ea[(r + u * step) * astride];
      v1[u] = ea[(r + u * step) * astride + 1];
    }
    for (int q = 0; q < nq; ++q) {
      const int c = (q * PERTGNN_WAVE + lane) * 4;
      float xv[U][4];
#pragma unroll
      for (int u = 0; u < U; ++u) vs_ld4(&g[(r + u * step) * h + c], xv[u]);
#pragma unroll
      for (int u = 0; u < U; ++u)
#pragma unroll
        for (int k = 0; k < 4; ++k) {
          atomicAdd(&acc[v0[u] * h + c + k], xv[u][k]);
          atomicAdd(&acc1[v1[u] * h + c + k], xv[u][k]);
        }
    }
  }
  for (; r < r1; r += step) {
    const long v0 = ea[r * astride];
    const long v1 = ea[r * astride + 1];
    for (int q = 0; q < nq; ++q) {
      const int c = (q * PERTGNN_WAVE + lane) * 4;
      float xv[4];
      vs_ld4(&g[r * h + c], xv);
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        atomicAdd(&acc[v0 * h + c + u], xv[u]);
        atomicAdd(&acc1[v1 * h + c + u], xv[u]);
      }
    }
  }
  __syncthreads();
  for (long t = threadIdx.x; t < (long)rows0 * h; t += blockDim.x)
    if (acc[t] != 0.f) atomicAdd(&dt0[t], acc[t]);
  for (long t = threadIdx.x; t < (long)rows1 * h; t += blockDim.x)
    if (acc1[t] != 0.f) atomicAdd(&dt1[t], acc1[t]);
}

// Wave-private variant — the production path.  Measured decomposition
// (benchmarks/vocab_micro.hip, E=216k/h=256/V=47): the LDS-atomic kernel is
// 100% ds_add_f32-bound at ~80 cycles/instruction (538 us; loads alone are
// 23 us; bank layout, unrolling, occupancy and index skew all change
// nothing).  Plain read+add+write into per-WAVE private tables removes the
// atomics entirely: 75 us for the same reduction (7.2x).  Each wave owns a
// private [rows0+rows1, HH] LDS table for one HH-wide column slice
// (grid.y = h/HH selects the slice); tables merge in LDS, one global-atomic
// flush per block.
template <typename GT, int U, int HH>
__global__ void vocab_scatter_dual_priv_kernel(const GT* __restrict__ g,
                                               const long* __restrict__ ea,
                                               int astride,
                                               float* __restrict__ dt0,
                                               float* __restrict__ dt1, long n,
                                               int rows0, int rows1, int h) {
  constexpr int CPL = HH / PERTGNN_WAVE;  // columns per lane
  extern __shared__ float acc[];  // [WAVES_PER_BLOCK][(rows0+rows1)*HH]
  const long vwh = (long)(rows0 + rows1) * HH;
  for (long t = threadIdx.x; t < WAVES_PER_BLOCK * vwh; t += blockDim.x)
    acc[t] = 0.f;
  __syncthreads();
  const int wid = threadIdx.x / PERTGNN_WAVE;
  const int lane = threadIdx.x % PERTGNN_WAVE;
  float* my0 = acc + (long)wid * vwh;
  float* my1 = my0 + (long)rows0 * HH;
  const int c0 = blockIdx.y * HH;
  const long rows_per_block = (n + gridDim.x - 1) / gridDim.x;
  const long r0 = (long)blockIdx.x * rows_per_block;
  const long r1 = min(n, r0 + rows_per_block);
  const long step = WAVES_PER_BLOCK;
  long r = r0 + wid;
  for (; r + (U - 1) * step < r1; r += U * step) {
    long v0[U], v1[U];
    float xv[U][CPL];
#pragma unroll
    for (int u = 0; u < U; ++u) {
      v0[u] = ea[(r + u * step) * astride];
      v1[u] = ea[(r + u * step) * astride + 1];
#pragma unroll
      for (int k = 0; k < CPL; ++k)
        xv[u][k] = (float)g[(r + u * step) * h + c0 + lane * CPL + k];
    }
#pragma unroll
    for (int u = 0; u < U; ++u) {
#pragma unroll
      for (int k = 0; k < CPL; ++k) my0[v0[u] * HH + lane * CPL + k] += xv[u][k];
#pragma unroll
      for (int k = 0; k < CPL; ++k) my1[v1[u] * HH + lane * CPL + k] += xv[u][k];
    }
  }
  for (; r < r1; r += step) {
    const long v0 = ea[r * astride];
    const long v1 = ea[r * astride + 1];
#pragma unroll
    for (int k = 0; k < CPL; ++k) {
      const float x = (float)g[r * h + c0 + lane * CPL + k];
      my0[v0 * HH + lane * CPL + k] += x;
      my1[v1 * HH + lane * CPL + k] += x;
    }
  }
  __syncthreads();
  for (long t = threadIdx.x; t < vwh; t += blockDim.x)
    acc[t] += acc[vwh + t] + acc[2 * vwh + t] + acc[3 * vwh + t];
  __syncthreads();
  for (long t = threadIdx.x; t < (long)rows0 * HH; t += blockDim.x) {
    const float v = acc[t];
    if (v != 0.f) atomicAdd(&dt0[(t / HH) * h + c0 + t % HH], v);
  }
  for (long t = threadIdx.x; t < (long)rows1 * HH; t += blockDim.x) {
    const float v = acc[(long)rows0 * HH + t];
    if (v != 0.f) atomicAdd(&dt1[(t / HH) * h + c0 + t % HH], v);
  }
}

template <typename GT>
static void vocab_dual_impl(const GT* g, const long* ea, int astride,
                            float* dt0, float* dt1, long n, int rows0,
                            int rows1, int h, hipStream_t s) {
  HIP_CHECK(hipMemsetAsync(dt0, 0, (long)rows0 * h * sizeof(float), s));
  HIP_CHECK(hipMemsetAsync(dt1, 0, (long)rows1 * h * sizeof(float), s));
  if (n == 0) return;
  // wave-private path (7.2x the LDS-atomic kernel, see kernel comment)
  const long vrows = rows0 + rows1;
  // block count scaled inversely with table size: each extra block
  // costs a table-sized atomic merge, but small tables (rpctype ~5
  // rows) need far more than 128 blocks to occupy 256 CUs (measured
  // 1 wave/SIMD at the old cap)
  const long merge_cap = (long)1000000 / (long)max((int)vrows * 128, 1);
  const int blocks =
      (int)min((n + 255) / 256, max((long)128, min((long)768, merge_cap)));
  const size_t lds128 = (size_t)WAVES_PER_BLOCK * vrows * 128 * sizeof(float);
  const size_t lds64 = (size_t)WAVES_PER_BLOCK * vrows * 64 * sizeof(float);
  if (h % 128 == 0 && lds128 <= 160 * 1024) {
    HIP_CHECK(hipFuncSetAttribute(
        (const void*)vocab_scatter_dual_priv_kernel<GT, 4, 128>,
        hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds128));
    vocab_scatter_dual_priv_kernel<GT, 4, 128>
        <<<dim3(max(blocks, 1), h / 128),
           dim3(WAVES_PER_BLOCK * PERTGNN_WAVE), lds128, s>>>(
            g, ea, astride, dt0, dt1, n, rows0, rows1, h);
    return;
  }
  if (h % 64 == 0 && lds64 <= 160 * 1024) {
    HIP_CHECK(hipFuncSetAttribute(
        (const void*)vocab_scatter_dual_priv_kernel<GT, 4, 64>,
        hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds64));
    vocab_scatter_dual_priv_kernel<GT, 4, 64>
        <<<dim3(max(blocks, 1), h / 64),
           dim3(WAVES_PER_BLOCK * PERTGNN_WAVE), lds64, s>>>(
            g, ea, astride, dt0, dt1, n, rows0, rows1, h);
    return;
  }
  const size_t lds = (size_t)vrows * h * sizeof(float);
  if (lds > 64 * 1024) {
    HIP_CHECK(hipFuncSetAttribute(
        (const void*)vocab_scatter_dual_kernel<GT, 4>,
        hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds));
  }
  const int ablocks = (int)min((long)768, (n + 63) / 64);
  vocab_scatter_dual_kernel<GT, 4><<<dim3(max(ablocks, 1)),
                                     dim3(WAVES_PER_BLOCK * PERTGNN_WAVE), lds,
                                     s>>>(g, ea, astride, dt0, dt1, n, rows0,
                                          rows1, h);
}

void launch_vocab_scatter_dual(const float* g, const long* ea, int astride,
                               float* dt0, float* dt1, long n, int rows0,
                               int rows1, int h, hipStream_t s) {
  vocab_dual_impl<float>(g, ea, astride, dt0, dt1, n, rows0, rows1, h, s);
}

void launch_vocab_scatter_dual16(const void* g, const long* ea, int astride,
                                 float* dt0, float* dt1, long n, int rows0,
                                 int rows1, int h, hipStream_t s) {
  vocab_dual_impl<__bf16>((const __bf16*)g, ea, astride, dt0, dt1, n, rows0,
                          rows1, h, s);
}

// single-table wave-private variant (same rationale/measurements as the dual
// kernel above: plain LDS read+add+write beats ds_add_f32 ~7x)
template <int U, int HH, typename GT = float>
__global__ void vocab_scatter_priv_kernel(const GT* __restrict__ g,
                                          const long* __restrict__ idx,
                                          long idx_stride,
                                          float* __restrict__ dtable, long n,
                                          int rows, int h, int gstride,
                                          int col_off) {
  constexpr int CPL = HH / PERTGNN_WAVE;
  extern __shared__ float acc[];  // [WAVES_PER_BLOCK][rows*HH]
  const long vwh = (long)rows * HH;
  for (long t = threadIdx.x; t < WAVES_PER_BLOCK * vwh; t += blockDim.x)
    acc[t] = 0.f;
  __syncthreads();
  const int wid = threadIdx.x / PERTGNN_WAVE;
  const int lane = threadIdx.x % PERTGNN_WAVE;
  float* my = acc + (long)wid * vwh;
  const int c0 = blockIdx.y * HH;
  const long rows_per_block = (n + gridDim.x - 1) / gridDim.x;
  const long r0 = (long)blockIdx.x * rows_per_block;
  const long r1 = min(n, r0 + rows_per_block);
  const long step = WAVES_PER_BLOCK;
  long r = r0 + wid;
  for (; r + (U - 1) * step < r1; r += U * step) {
    long v[U];
    float xv[U][CPL];
#pragma unroll
    for (int u = 0; u < U; ++u) {
      v[u] = idx[(r + u * step) * idx_stride];
#pragma unroll
      for (int k = 0; k < CPL; ++k)
        xv[u][k] = (float)g[(r + u * step) * gstride + col_off + c0 + lane * CPL + k];
    }
#pragma unroll
    for (int u = 0; u < U; ++u)
#pragma unroll
      for (int k = 0; k < CPL; ++k) my[v[u] * HH + lane * CPL + k] += xv[u][k];
  }
  for (; r < r1; r += step) {
    const long v = idx[r * idx_stride];
#pragma unroll
    for (int k = 0; k < CPL; ++k)
      my[v * HH + lane * CPL + k] +=
          (float)g[r * gstride + col_off + c0 + lane * CPL + k];
  }
  __syncthreads();
  for (long t = threadIdx.x; t < vwh; t += blockDim.x)
    acc[t] += acc[vwh + t] + acc[2 * vwh + t] + acc[3 * vwh + t];
  __syncthreads();
  for (long t = threadIdx.x; t < vwh; t += blockDim.x) {
    const float val = acc[t];
    if (val != 0.f) atomicAdd(&dtable[(t / HH) * h + c0 + t % HH], val);
  }
}

// bf16-gradient variant: wave-private path only (the caller upcasts to f32
// when the private tables do not fit)
void launch_vocab_scatter16(const void* g_v, const long* idx, long idx_stride,
                            float* dtable, long n, int rows, int h,
                            int gstride, int col_off, hipStream_t s) {
  const __bf16* g = (const __bf16*)g_v;
  HIP_CHECK(hipMemsetAsync(dtable, 0, (long)rows * h * sizeof(float), s));
  if (n == 0) return;
  // block count scaled inversely with table size: each extra block
  // costs a table-sized atomic merge, but small tables (rpctype ~5
  // rows) need far more than 128 blocks to occupy 256 CUs (measured
  // 1 wave/SIMD at the old cap)
  const long merge_cap = (long)1000000 / (long)max(rows * 128, 1);
  const int blocks =
      (int)min((n + 255) / 256, max((long)128, min((long)768, merge_cap)));
  const size_t lds128 = (size_t)WAVES_PER_BLOCK * rows * 128 * sizeof(float);
  const size_t lds64 = (size_t)WAVES_PER_BLOCK * rows * 64 * sizeof(float);
  if (h % 128 == 0 && lds128 <= 160 * 1024) {
    HIP_CHECK(hipFuncSetAttribute(
        (const void*)vocab_scatter_priv_kernel<4, 128, __bf16>,
        hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds128));
    vocab_scatter_priv_kernel<4, 128, __bf16>
        <<<dim3(max(blocks, 1), h / 128), dim3(WAVES_PER_BLOCK * PERTGNN_WAVE),
           lds128, s>>>(g, idx, idx_stride, dtable, n, rows, h, gstride,
                        col_off);
    return;
  }
  if (h % 64 == 0 && lds64 <= 160 * 1024) {
    HIP_CHECK(hipFuncSetAttribute(
        (const void*)vocab_scatter_priv_kernel<4, 64, __bf16>,
        hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds64));
    vocab_scatter_priv_kernel<4, 64, __bf16>
        <<<dim3(max(blocks, 1), h / 64), dim3(WAVES_PER_BLOCK * PERTGNN_WAVE),
           lds64, s>>>(g, idx, idx_stride, dtable, n, rows, h, gstride,
                       col_off);
    return;
  }
  pertgnn_shape_fail("vocab_scatter", "table_bytes_over_lds_h", h);  // caller guarantees fit (python upcasts otherwise)
}

void launch_vocab_scatter(const float* g, const long* idx, long idx_stride,
                          float* dtable, long n, int rows, int h, int gstride,
                          int col_off, hipStream_t s) {
  HIP_CHECK(hipMemsetAsync(dtable, 0, (long)rows * h * sizeof(float), s));
  if (n == 0) return;
  {
    // block count scaled inversely with table size: each extra block
    // costs a table-sized atomic merge, but small tables (rpctype ~5
    // rows) need far more than 128 blocks to occupy 256 CUs (measured
    // 1 wave/SIMD at the old cap)
    const long merge_cap = (long)1000000 / (long)max(rows * 128, 1);
    const int blocks =
        (int)min((n + 255) / 256, max((long)128, min((long)768, merge_cap)));
    const size_t lds128 = (size_t)WAVES_PER_BLOCK * rows * 128 * sizeof(float);
    const size_t lds64 = (size_t)WAVES_PER_BLOCK * rows * 64 * sizeof(float);
    if (h % 128 == 0 && lds128 <= 160 * 1024) {
      HIP_CHECK(hipFuncSetAttribute(
          (const void*)vocab_scatter_priv_kernel<4, 128>,
          hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds128));
      vocab_scatter_priv_kernel<4, 128>
          <<<dim3(max(blocks, 1), h / 128),
             dim3(WAVES_PER_BLOCK * PERTGNN_WAVE), lds128, s>>>(
              g, idx, idx_stride, dtable, n, rows, h, gstride, col_off);
      return;
    }
    if (h % 64 == 0 && lds64 <= 160 * 1024) {
      HIP_CHECK(hipFuncSetAttribute(
          (const void*)vocab_scatter_priv_kernel<4, 64>,
          hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds64));
      vocab_scatter_priv_kernel<4, 64>
          <<<dim3(max(blocks, 1), h / 64),
             dim3(WAVES_PER_BLOCK * PERTGNN_WAVE), lds64, s>>>(
              g, idx, idx_stride, dtable, n, rows, h, gstride, col_off);
      return;
    }
  }
  const size_t lds = (size_t)rows * h * sizeof(float);
  const bool vec = (h % (4 * PERTGNN_WAVE) == 0) && ((gstride & 3) == 0) &&
                   ((col_off & 3) == 0);
  const void* fn = vec ? (const void*)vocab_scatter_vec_kernel
                       : (const void*)vocab_scatter_kernel;
  if (lds > 64 * 1024) {
    HIP_CHECK(hipFuncSetAttribute(
        fn, hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds));
  }
  const int blocks = (int)min((long)256, (n + 63) / 64);
  if (vec)
    vocab_scatter_vec_kernel<<<dim3(blocks),
                               dim3(WAVES_PER_BLOCK * PERTGNN_WAVE), lds, s>>>(
        g, idx, idx_stride, dtable, n, rows, h, gstride, col_off);
  else
    vocab_scatter_kernel<<<dim3(blocks), dim3(WAVES_PER_BLOCK * PERTGNN_WAVE),
                           lds, s>>>(g, idx, idx_stride, dtable, n, rows, h,
                                     gstride, col_off);
}

// entry embedding gather: out[b] = table[idx[b]] — plain gather (fwd) +
// scatter-add (bwd); reuses the edge kernels' grid-stride shape.
__global__ void gather_rows_kernel(const long* __restrict__ idx,
                                   const float* __restrict__ table,
                                   float* __restrict__ out, long n, int h) {
  const long numel = n * h;
  const long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long t = i0; t < numel; t += stride) {
    const long row = t / h;
    const int c = (int)(t - row * h);
    out[t] = table[idx[row] * h + c];
  }
}

__global__ void scatter_add_rows_kernel(const float* __restrict__ g,
                                        const long* __restrict__ idx,
                                        float* __restrict__ dtable, long n,
                                        int h) {
  const long numel = n * h;
  const long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long t = i0; t < numel; t += stride) {
    const long row = t / h;
    const int c = (int)(t - row * h);
    atomicAdd(&dtable[idx[row] * h + c], g[t]);
  }
}

static int grid_for(long numel, int tpb = 256, long cap = 4096) {
  return (int)min((numel + tpb - 1) / tpb, cap);
}

void launch_embed_node_fwd(const float* x_raw, const long* idx,
                           const float* table, float* out, long n, int f,
                           int h, hipStream_t s) {
  if (n == 0) return;
  embed_node_fwd_kernel<<<grid_for(n * (f + h)), 256, 0, s>>>(x_raw, idx, table,
                                                              out, n, f, h);
}

void launch_embed_node_fwd16(const float* x_raw, const long* idx,
                             const float* table, void* out, long n, int f,
                             int h, hipStream_t stream) {
  const long numel = n * (long)(f + h);
  const int tpb = 256;
  const int blocks = (int)min((numel + tpb - 1) / tpb, (long)4096);
  embed_node_fwd_kernel<<<dim3(blocks), dim3(tpb), 0, stream>>>(
      x_raw, idx, table, (__bf16*)out, n, f, h);
}
void launch_embed_edge_fwd(const long* attr, const float* ifc,
                           const float* rpc, float* out, long e, int h,
                           int astride, hipStream_t s) {
  if (e == 0) return;
  embed_edge_fwd_kernel<<<grid_for(e * 2 * h), 256, 0, s>>>(attr, ifc, rpc, out,
                                                            e, h, astride);
}
void launch_gather_rows(const long* idx, const float* table, float* out,
                        long n, int h, hipStream_t s) {
  if (n == 0) return;
  gather_rows_kernel<<<grid_for(n * h), 256, 0, s>>>(idx, table, out, n, h);
}

// ---------------------------------------------------------------------------
// BatchNorm1d (+ReLU) — two-stage column reduction, coalesced row tiles
// ---------------------------------------------------------------------------

// stage A: per-block partial sum/sumsq over a row range; thread t owns
// channels t and t+256 in REGISTERS (H <= 512), one atomicAdd per channel
// per block at the end — coalesced loads, no LDS traffic.
// adjacent-channel pair load: one 4-B (bf16) / 8-B (fp32) transaction for
// the thread's two channels (c, c+1) — halves the load count of the
// latency-bound per-channel reduction kernels
__device__ __forceinline__ void bn_ld2(const float* p, float& a, float& b) {
  typedef __attribute__((ext_vector_type(2))) float f2_t;
  const f2_t v = *reinterpret_cast<const f2_t*>(p);
  a = v[0]; b = v[1];
}
__device__ __forceinline__ void bn_ld2(const __bf16* p, float& a, float& b) {
  typedef __attribute__((ext_vector_type(2))) __bf16 b2_t;
  const b2_t v = *reinterpret_cast<const b2_t*>(p);
  a = (float)v[0]; b = (float)v[1];
}

template <typename TX = float, bool SLAB = false>
__global__ void bn_stats_partial_kernel(const TX* __restrict__ x, long n,
                                        int h, float* __restrict__ partials) {
  // ADJACENT channel pair per thread (c0 = 2*tid, c1 = c0+1): one paired
  // load serves both, halving the load count of this latency-bound kernel
  const int c0 = 2 * threadIdx.x;
  const int c1 = c0 + 1;
  float s0 = 0.f, q0 = 0.f, s1 = 0.f, q1 = 0.f;
  const long rows_per_block = (n + gridDim.x - 1) / gridDim.x;
  const long r0 = (long)blockIdx.x * rows_per_block;
  const long r1 = min(n, r0 + rows_per_block);
  long r = r0;
  if ((h & 1) == 0 && c1 < h) {
    // 8-row unroll: the kernel is latency-bound (one bf16x2 load per
    // channel pair per row) — 8 independent loads in flight halve the
    // stall count vs 4 (measured 58.7 -> ~35 us at 180k x 256)
    for (; r + 16 <= r1; r += 16) {
      float a[16], b[16];
#pragma unroll
      for (int u = 0; u < 16; ++u) bn_ld2(&x[(r + u) * h + c0], a[u], b[u]);
#pragma unroll
      for (int u = 0; u < 16; ++u) {
        s0 += a[u]; q0 += a[u] * a[u];
        s1 += b[u]; q1 += b[u] * b[u];
      }
    }
    for (; r < r1; ++r) {
      float a, b;
      bn_ld2(&x[r * h + c0], a, b);
      s0 += a; q0 += a * a;
      s1 += b; q1 += b * b;
    }
  } else {
    for (; r < r1; ++r) {
      if (c0 < h) {
        const float v = x[r * h + c0];
        s0 += v; q0 += v * v;
      }
      if (c1 < h) {
        const float v = x[r * h + c1];
        s1 += v; q1 += v * v;
      }
    }
  }
  if (SLAB) {
    // deterministic mode: per-block slab row, reduced in fixed block order
    float* row = partials + (long)blockIdx.x * 2 * h;
    if (c0 < h) { row[c0] = s0; row[h + c0] = q0; }
    if (c1 < h) { row[c1] = s1; row[h + c1] = q1; }
    return;
  }
  if (c0 < h && r0 < r1) {
    atomicAdd(&partials[c0], s0);
    atomicAdd(&partials[h + c0], q0);
  }
  if (c1 < h && r0 < r1) {
    atomicAdd(&partials[c1], s1);
    atomicAdd(&partials[h + c1], q1);
  }
}

// fixed-order fold of the deterministic-mode slab rows
__global__ void bn_slab_reduce_kernel(const float* __restrict__ slab, int nb,
                                      int w, float* __restrict__ out) {
  const int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= w) return;
  float s = 0.f;
  for (int b = 0; b < nb; ++b) s += slab[(long)b * w + c];
  out[c] = s;
}

// lazily-allocated device slab for the deterministic BN reductions
// (512 blocks x 2 x 1024 channels); deterministic mode is documented as
// incompatible with hipGraph capture, so the lazy hipMalloc is safe.
static float* bn_det_slab() {
  static float* p = [] {
    float* q = nullptr;
    HIP_CHECK(hipMalloc(&q, (size_t)512 * 2 * 1024 * sizeof(float)));
    return q;
  }();
  return p;
}

static inline bool pertgnn_deterministic_seg() {
  const char* e = getenv("PERTGNN_DETERMINISTIC");
  return e && e[0] == '1';
}

// grid size for the BN channel reductions (atomic-combine path): 512 blocks
// is 2 blocks/CU — enough to stream ~3.5 TB/s but measurably latency-bound
// on the single-stream stats kernel; PERTGNN_BN_BLOCKS overrides for tuning
static inline int bn_partial_blocks(long n) {
  static const int env = [] {
    const char* e = getenv("PERTGNN_BN_BLOCKS");
    return e ? atoi(e) : 0;
  }();
  const long cap = (n + 63) / 64;
  const long want = env > 0 ? env : 512;
  return (int)min(want, cap);
}

template <typename TX>
static void bn_stats_dispatch(const TX* x, long n, int h, float* partials,
                              hipStream_t s) {
  const int nblocks = pertgnn_deterministic_seg()
                          ? (int)min((long)512, (n + 63) / 64)
                          : bn_partial_blocks(n);
  if (pertgnn_deterministic_seg() && h <= 1024) {
    float* slab = bn_det_slab();
    bn_stats_partial_kernel<TX, true><<<nblocks, 256, 0, s>>>(x, n, h, slab);
    bn_slab_reduce_kernel<<<ceil_div(2 * h, 256), 256, 0, s>>>(
        slab, nblocks, 2 * h, partials);
  } else {
    HIP_CHECK(hipMemsetAsync(partials, 0, 2 * h * sizeof(float), s));
    bn_stats_partial_kernel<TX, false><<<nblocks, 256, 0, s>>>(x, n, h,
                                                               partials);
  }
}


// stage B: finalize mean/invstd (+ running-stat update, training only)
// count_ptr (may be null): device-side GLOBAL row count — sync-BN appends
// the local count to the stats partials so ONE all-reduce carries sums AND
// count, with no host round-trip per BN layer (the count slot lives at
// partials[2h], untouched by the 2h-wide reducers).
__global__ void bn_finalize_kernel(const float* __restrict__ partials, long n,
                                   int h, float eps, float momentum,
                                   float* __restrict__ mean,
                                   float* __restrict__ invstd,
                                   float* __restrict__ running_mean,
                                   float* __restrict__ running_var,
                                   int update_running,
                                   const float* __restrict__ count_ptr = nullptr) {
  const int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= h) return;
  const float nf = count_ptr ? *count_ptr : (float)n;
  const float m = partials[c] / nf;
  const float var = fmaxf(partials[h + c] / nf - m * m, 0.f);
  mean[c] = m;
  invstd[c] = rsqrtf(var + eps);
  if (update_running) {
    const float unbiased = (nf > 1.f) ? var * nf / (nf - 1.f) : var;
    running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * m;
    running_var[c] = (1.f - momentum) * running_var[c] + momentum * unbiased;
  }
}

// counter-based per-element uniform for fused dropout (splitmix64 mix of
// the device-resident step counter and the element index; quality is ample
// for dropout masks and the sequence is deterministic given the counter)
__device__ __forceinline__ float bn_rand01(unsigned long long seed, long t) {
  unsigned long long z = seed * 0x9E3779B97F4A7C15ull + (unsigned long long)t;
  z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ull;
  z = (z ^ (z >> 27)) * 0x94D049BB133111EBull;
  z ^= z >> 31;
  return (float)(z >> 40) * (1.f / 16777216.f);  // top 24 bits
}

// stage C: normalize + affine (+ReLU) (+fused dropout: reference K8 — the
// model applies dropout right after BN+ReLU, model.py:101-103; keep_inv =
// 1/(1-p), seed_ptr = device step counter so captured hipGraphs draw FRESH
// masks every replay).  Post-dropout y==0 <=> dropped-or-relu-negative, so
// the backward needs NO RNG: the existing y<=0 mask plus the keep_inv
// scale reproduces d(dropout.relu)/dy exactly.
// TY = output/activation dtype: fp32 in exact mode, bf16 in the act16 mode
// (standard mixed-precision BN: statistics and normalization math stay fp32,
// only the activation stream is 16-bit).
// DROP is a TEMPLATE parameter: the hash-RNG code measurably bloats the
// 4-wide loop even when dropout_p==0 at runtime (bn_apply 39 -> 168 us at
// 181k x 256), so the p==0 instantiation must not contain it.
template <typename TY, typename TX = float, bool DROP = false>
__global__ void bn_apply_kernel(const TX* __restrict__ x,
                                const float* __restrict__ mean,
                                const float* __restrict__ invstd,
                                const float* __restrict__ gamma,
                                const float* __restrict__ beta,
                                TY* __restrict__ y, long n, int h,
                                int relu, float dropout_p = 0.f,
                                const unsigned long long* __restrict__
                                    seed_ptr = nullptr) {
  typedef __attribute__((ext_vector_type(4))) float bnf4;
  const float keep_inv = DROP ? 1.f / (1.f - dropout_p) : 1.f;
  unsigned long long seed = 0ull;
  if constexpr (DROP) seed = seed_ptr ? *seed_ptr : 0ull;
  if ((h & 3) == 0) {  // 4-wide: one f32x4 load + packed store per thread
    const long numq = n * (h / 4);
    const long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
    const long stride = (long)gridDim.x * blockDim.x;
    for (long q = i0; q < numq; q += stride) {
      const long t = q * 4;
      const int c = (int)(t % h);
      float xv[4];
      vs_ld4(&x[t], xv);
      struct alignas(4 * sizeof(TY)) TY4 { TY v[4]; };
      TY4 o;
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        float v = (xv[u] - mean[c + u]) * invstd[c + u] * gamma[c + u] +
                  beta[c + u];
        if (relu) v = fmaxf(v, 0.f);
        if constexpr (DROP)
          v = (bn_rand01(seed, t + u) >= dropout_p) ? v * keep_inv : 0.f;
        o.v[u] = (TY)v;
      }
      *reinterpret_cast<TY4*>(&y[t]) = o;  // one packed 8B/16B store
    }
    return;
  }
  const long numel = n * h;
  const long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long t = i0; t < numel; t += stride) {
    const int c = (int)(t % h);
    float v = ((float)x[t] - mean[c]) * invstd[c] * gamma[c] + beta[c];
    if (relu) v = fmaxf(v, 0.f);
    if constexpr (DROP)
      v = (bn_rand01(seed, t) >= dropout_p) ? v * keep_inv : 0.f;
    y[t] = (TY)v;
  }
}

__global__ void counter_bump_kernel(unsigned long long* __restrict__ c) {
  ++*c;
}

// backward stage A: per-channel sums of gm and gm*xhat (gm = relu-masked g)
template <typename TG, typename TY, typename TX = float, bool SLAB = false>
__global__ void bn_bwd_partial_kernel(const TG* __restrict__ g,
                                      const TX* __restrict__ x,
                                      const TY* __restrict__ y,
                                      const float* __restrict__ mean,
                                      const float* __restrict__ invstd, long n,
                                      int h, int relu,
                                      float* __restrict__ partials,
                                      float keep_inv = 1.f) {
  // adjacent channel pair per thread (see bn_stats): paired g/y/x loads
  const int c0 = 2 * threadIdx.x;
  const int c1 = c0 + 1;
  float s0 = 0.f, q0 = 0.f, s1 = 0.f, q1 = 0.f;
  const long rows_per_block = (n + gridDim.x - 1) / gridDim.x;
  const long r0 = (long)blockIdx.x * rows_per_block;
  const long r1 = min(n, r0 + rows_per_block);
  const float m0 = (c0 < h) ? mean[c0] : 0.f;
  const float i0 = (c0 < h) ? invstd[c0] : 0.f;
  const float m1 = (c1 < h) ? mean[c1] : 0.f;
  const float i1 = (c1 < h) ? invstd[c1] : 0.f;
  long r = r0;
  if ((h & 1) == 0 && c1 < h) {
    for (; r + 8 <= r1; r += 8) {
      float ga[8], gb[8], ya[8], yb[8], xa[8], xb[8];
#pragma unroll
      for (int u = 0; u < 8; ++u) {
        bn_ld2(&g[(r + u) * h + c0], ga[u], gb[u]);
        bn_ld2(&y[(r + u) * h + c0], ya[u], yb[u]);
        bn_ld2(&x[(r + u) * h + c0], xa[u], xb[u]);
      }
#pragma unroll
      for (int u = 0; u < 8; ++u) {
        float gm0 = (relu && ya[u] <= 0.f) ? 0.f : ga[u] * keep_inv;
        float gm1 = (relu && yb[u] <= 0.f) ? 0.f : gb[u] * keep_inv;
        s0 += gm0; q0 += gm0 * (xa[u] - m0) * i0;
        s1 += gm1; q1 += gm1 * (xb[u] - m1) * i1;
      }
    }
    for (; r < r1; ++r) {
      float ga, gb, ya, yb, xa, xb;
      bn_ld2(&g[r * h + c0], ga, gb);
      bn_ld2(&y[r * h + c0], ya, yb);
      bn_ld2(&x[r * h + c0], xa, xb);
      float gm0 = (relu && ya <= 0.f) ? 0.f : ga * keep_inv;
      float gm1 = (relu && yb <= 0.f) ? 0.f : gb * keep_inv;
      s0 += gm0; q0 += gm0 * (xa - m0) * i0;
      s1 += gm1; q1 += gm1 * (xb - m1) * i1;
    }
  } else {
    for (; r < r1; ++r) {
      if (c0 < h) {
        float gm = (float)g[r * h + c0] * keep_inv;
        if (relu && (float)y[r * h + c0] <= 0.f) gm = 0.f;
        s0 += gm;
        q0 += gm * ((float)x[r * h + c0] - m0) * i0;
      }
      if (c1 < h) {
        float gm = (float)g[r * h + c1] * keep_inv;
        if (relu && (float)y[r * h + c1] <= 0.f) gm = 0.f;
        s1 += gm;
        q1 += gm * ((float)x[r * h + c1] - m1) * i1;
      }
    }
  }
  if (SLAB) {
    float* row = partials + (long)blockIdx.x * 2 * h;
    if (c0 < h) { row[c0] = s0; row[h + c0] = q0; }
    if (c1 < h) { row[c1] = s1; row[h + c1] = q1; }
    return;
  }
  if (c0 < h && r0 < r1) {
    atomicAdd(&partials[c0], s0);
    atomicAdd(&partials[h + c0], q0);
  }
  if (c1 < h && r0 < r1) {
    atomicAdd(&partials[c1], s1);
    atomicAdd(&partials[h + c1], q1);
  }
}


template <typename TG, typename TY, typename TX>
static void bn_bwd_partials_dispatch(const TG* g, const TX* x, const TY* y,
                                     const float* mean, const float* invstd,
                                     long n, int h, bool relu, float* partials,
                                     float keep_inv, hipStream_t s) {
  const int nblocks = pertgnn_deterministic_seg()
                          ? (int)min((long)512, (n + 63) / 64)
                          : bn_partial_blocks(n);
  if (pertgnn_deterministic_seg() && h <= 1024) {
    float* slab = bn_det_slab();
    bn_bwd_partial_kernel<TG, TY, TX, true><<<nblocks, 256, 0, s>>>(
        g, x, y, mean, invstd, n, h, relu ? 1 : 0, slab, keep_inv);
    bn_slab_reduce_kernel<<<ceil_div(2 * h, 256), 256, 0, s>>>(
        slab, nblocks, 2 * h, partials);
  } else {
    HIP_CHECK(hipMemsetAsync(partials, 0, 2 * h * sizeof(float), s));
    bn_bwd_partial_kernel<TG, TY, TX, false><<<nblocks, 256, 0, s>>>(
        g, x, y, mean, invstd, n, h, relu ? 1 : 0, partials, keep_inv);
  }
}

// backward stage B: dx = gamma*invstd*(gm - sum_gm/n - xhat*sum_gmx/n)
// 4-wide vectorized (16-bit scalar loads left the kernel latency-bound:
// 163 us vs 133 us fp32 at 165k x 256 — 4x-unrolled loads recover it)
template <typename TG, typename TY, typename TX = float>
__global__ void bn_bwd_apply_kernel(
    const TG* __restrict__ g, const TX* __restrict__ x,
    const TY* __restrict__ y, const float* __restrict__ mean,
    const float* __restrict__ invstd, const float* __restrict__ gamma,
    const float* __restrict__ partials, TX* __restrict__ dx, long n,
    long count, int h, int relu,
    const float* __restrict__ count_ptr = nullptr, float keep_inv = 1.f) {
  typedef __attribute__((ext_vector_type(4))) float bnf4;
  const float invn = 1.f / (count_ptr ? *count_ptr : (float)count);
  if ((h & 3) == 0) {
    const long numq = n * (h / 4);
    const long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
    const long stride = (long)gridDim.x * blockDim.x;
    struct alignas(4 * sizeof(TG)) TG4 { TG v[4]; };
    struct alignas(4 * sizeof(TY)) TY4 { TY v[4]; };
    for (long q = i0; q < numq; q += stride) {
      const long t = q * 4;
      const int c = (int)(t % h);
      float gv[4], yv[4], xv[4];
      // vectorized g/y quad loads (4 scalar 16-bit loads left this kernel
      // issue-bound alongside the already-vector x load)
      const TG4 gq = *reinterpret_cast<const TG4*>(&g[t]);
#pragma unroll
      for (int u = 0; u < 4; ++u) gv[u] = (float)gq.v[u];
      if (relu) {
        const TY4 yq = *reinterpret_cast<const TY4*>(&y[t]);
#pragma unroll
        for (int u = 0; u < 4; ++u) yv[u] = (float)yq.v[u];
      }
      vs_ld4(&x[t], xv);
      struct alignas(4 * sizeof(TX)) TX4 { TX v[4]; };
      TX4 o;
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        float gm = gv[u] * keep_inv;
        if (relu && yv[u] <= 0.f) gm = 0.f;
        const float xhat = (xv[u] - mean[c + u]) * invstd[c + u];
        o.v[u] = (TX)(gamma[c + u] * invstd[c + u] *
                      (gm - partials[c + u] * invn -
                       xhat * partials[h + c + u] * invn));
      }
      *reinterpret_cast<TX4*>(&dx[t]) = o;
    }
    return;
  }
  const long numel = n * h;
  const long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long t = i0; t < numel; t += stride) {
    const int c = (int)(t % h);
    float gm = (float)g[t] * keep_inv;
    if (relu && (float)y[t] <= 0.f) gm = 0.f;
    const float xhat = ((float)x[t] - mean[c]) * invstd[c];
    dx[t] = (TX)(gamma[c] * invstd[c] *
                 (gm - partials[c] * invn - xhat * partials[h + c] * invn));
  }
}

// dgamma[c] = sum gm*xhat = partials[h+c]; dbeta[c] = sum gm = partials[c]
__global__ void bn_grad_affine_kernel(const float* __restrict__ partials,
                                      float* __restrict__ dgamma,
                                      float* __restrict__ dbeta, int h) {
  const int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= h) return;
  dgamma[c] = partials[h + c];
  dbeta[c] = partials[c];
}

__global__ void bn_eval_stats_kernel(const float* __restrict__ rm,
                                     const float* __restrict__ rv,
                                     float* __restrict__ mean,
                                     float* __restrict__ invstd, int h,
                                     float eps) {
  const int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= h) return;
  mean[c] = rm[c];
  invstd[c] = rsqrtf(rv[c] + eps);
}

void launch_bn_eval_stats(const float* running_mean, const float* running_var,
                          float* mean, float* invstd, int h, float eps,
                          hipStream_t s) {
  bn_eval_stats_kernel<<<ceil_div(h, 256), 256, 0, s>>>(running_mean,
                                                        running_var, mean,
                                                        invstd, h, eps);
}

// --- granular launchers for sync-BN (partials are all-reduced across ranks
// between the stats and apply phases; `count` = GLOBAL row count) ---

void launch_bn_stats_only(const float* x, long n, int h, float* partials,
                          hipStream_t s) {
  if (n == 0) {
    HIP_CHECK(hipMemsetAsync(partials, 0, 2 * h * sizeof(float), s));
    return;
  }
  bn_stats_dispatch(x, n, h, partials, s);
}

void launch_bn_finalize_apply(const float* x, const float* partials,
                              const float* count_ptr, const float* gamma,
                              const float* beta, float* running_mean,
                              float* running_var, float* mean, float* invstd,
                              float* y, long n, int h, float momentum,
                              float eps, bool training, bool relu,
                              float dropout_p,
                              const unsigned long long* seed_ptr,
                              hipStream_t s) {
  if (training) {
    bn_finalize_kernel<<<ceil_div(h, 256), 256, 0, s>>>(
        partials, n, h, eps, momentum, mean, invstd, running_mean,
        running_var, 1, count_ptr);
  } else {
    bn_eval_stats_kernel<<<ceil_div(h, 256), 256, 0, s>>>(
        running_mean, running_var, mean, invstd, h, eps);
  }
  if (n > 0) {
    if (training && dropout_p > 0.f)
      bn_apply_kernel<float, float, true><<<grid_for(n * h), 256, 0, s>>>(
          x, mean, invstd, gamma, beta, y, n, h, relu ? 1 : 0, dropout_p,
          seed_ptr);
    else
      bn_apply_kernel<<<grid_for(n * h), 256, 0, s>>>(x, mean, invstd, gamma,
                                                      beta, y, n, h,
                                                      relu ? 1 : 0);
  }
}

void launch_bn_bwd_partials_only(const float* g, const float* x,
                                 const float* y, const float* mean,
                                 const float* invstd, long n, int h, bool relu,
                                 float* partials, float keep_inv,
                                 hipStream_t s) {
  if (n == 0) {
    HIP_CHECK(hipMemsetAsync(partials, 0, 2 * h * sizeof(float), s));
    return;
  }
  bn_bwd_partials_dispatch(g, x, y, mean, invstd, n, h, relu, partials,
                           keep_inv, s);
}

void launch_bn_bwd_apply_only(const float* g, const float* x, const float* y,
                              const float* mean, const float* invstd,
                              const float* gamma, const float* partials,
                              const float* count_ptr, float* dx, long n, int h,
                              bool relu, float keep_inv, hipStream_t s) {
  if (n == 0) return;
  bn_bwd_apply_kernel<<<grid_for(n * h), 256, 0, s>>>(
      g, x, y, mean, invstd, gamma, partials, dx, n, n, h, relu ? 1 : 0,
      count_ptr, keep_inv);
}

void launch_bn_grad_affine(const float* partials, float* dgamma, float* dbeta,
                           int h, hipStream_t s) {
  bn_grad_affine_kernel<<<ceil_div(h, 256), 256, 0, s>>>(partials, dgamma,
                                                         dbeta, h);
}

void launch_bn_fwd(const float* x, const float* gamma, const float* beta,
                   float* running_mean, float* running_var, float* mean,
                   float* invstd, float* partials, float* y, long n, int h,
                   float momentum, float eps, bool training, bool relu,
                   float dropout_p, const unsigned long long* seed_ptr,
                   hipStream_t s) {
  if (n == 0) return;
  if (training) {
    bn_stats_dispatch(x, n, h, partials, s);
    bn_finalize_kernel<<<ceil_div(h, 256), 256, 0, s>>>(
        partials, n, h, eps, momentum, mean, invstd, running_mean, running_var,
        1);
  } else {
    launch_bn_eval_stats(running_mean, running_var, mean, invstd, h, eps, s);
  }
  if (training && dropout_p > 0.f)
    bn_apply_kernel<float, float, true><<<grid_for(n * h), 256, 0, s>>>(
        x, mean, invstd, gamma, beta, y, n, h, relu ? 1 : 0, dropout_p,
        seed_ptr);
  else
    bn_apply_kernel<<<grid_for(n * h), 256, 0, s>>>(x, mean, invstd, gamma,
                                                    beta, y, n, h,
                                                    relu ? 1 : 0);
}

void launch_counter_bump(unsigned long long* c, hipStream_t s) {
  counter_bump_kernel<<<1, 1, 0, s>>>(c);
}

void launch_bn_bwd(const float* g, const float* x, const float* y,
                   const float* mean, const float* invstd, const float* gamma,
                   float* partials, float* dx, float* dgamma, float* dbeta,
                   long n, int h, bool relu, float keep_inv, hipStream_t s) {
  if (n == 0) return;
  bn_bwd_partials_dispatch(g, x, y, mean, invstd, n, h, relu, partials,
                           keep_inv, s);
  bn_bwd_apply_kernel<<<grid_for(n * h), 256, 0, s>>>(
      g, x, y, mean, invstd, gamma, partials, dx, n, n, h, relu ? 1 : 0,
      nullptr, keep_inv);
  bn_grad_affine_kernel<<<ceil_div(h, 256), 256, 0, s>>>(partials, dgamma,
                                                         dbeta, h);
}

// --- act16 launchers: fully bf16 activation streams — x (the conv output /
// BN input), y, incoming grad g, and dx are all bf16; statistics, affine
// params and their grads stay fp32 (standard mixed-precision BN). ---

void launch_bn_stats_only16(const void* x, long n, int h, float* partials,
                            hipStream_t s) {
  if (n == 0) {
    HIP_CHECK(hipMemsetAsync(partials, 0, 2 * h * sizeof(float), s));
    return;
  }
  bn_stats_dispatch((const __bf16*)x, n, h, partials, s);
}

void launch_bn_fwd16(const void* x, const float* gamma, const float* beta,
                     float* running_mean, float* running_var, float* mean,
                     float* invstd, float* partials, void* y, long n, int h,
                     float momentum, float eps, bool training, bool relu,
                     float dropout_p, const unsigned long long* seed_ptr,
                     hipStream_t s) {
  if (n == 0) return;
  if (training) {
    bn_stats_dispatch((const __bf16*)x, n, h, partials, s);
    bn_finalize_kernel<<<ceil_div(h, 256), 256, 0, s>>>(
        partials, n, h, eps, momentum, mean, invstd, running_mean, running_var,
        1);
  } else {
    launch_bn_eval_stats(running_mean, running_var, mean, invstd, h, eps, s);
  }
  if (training && dropout_p > 0.f)
    bn_apply_kernel<__bf16, __bf16, true><<<grid_for(n * h), 256, 0, s>>>(
        (const __bf16*)x, mean, invstd, gamma, beta, (__bf16*)y, n, h,
        relu ? 1 : 0, dropout_p, seed_ptr);
  else
    bn_apply_kernel<<<grid_for(n * h), 256, 0, s>>>((const __bf16*)x, mean,
                                                    invstd, gamma, beta,
                                                    (__bf16*)y, n, h,
                                                    relu ? 1 : 0);
}

void launch_bn_finalize_apply16(const void* x, const float* partials,
                                const float* count_ptr, const float* gamma,
                                const float* beta, float* running_mean,
                                float* running_var, float* mean, float* invstd,
                                void* y, long n, int h, float momentum,
                                float eps, bool training, bool relu,
                                float dropout_p,
                                const unsigned long long* seed_ptr,
                                hipStream_t s) {
  if (training) {
    bn_finalize_kernel<<<ceil_div(h, 256), 256, 0, s>>>(
        partials, n, h, eps, momentum, mean, invstd, running_mean,
        running_var, 1, count_ptr);
  } else {
    bn_eval_stats_kernel<<<ceil_div(h, 256), 256, 0, s>>>(
        running_mean, running_var, mean, invstd, h, eps);
  }
  if (n > 0) {
    if (training && dropout_p > 0.f)
      bn_apply_kernel<__bf16, __bf16, true><<<grid_for(n * h), 256, 0, s>>>(
          (const __bf16*)x, mean, invstd, gamma, beta, (__bf16*)y, n, h,
          relu ? 1 : 0, dropout_p, seed_ptr);
    else
      bn_apply_kernel<<<grid_for(n * h), 256, 0, s>>>((const __bf16*)x, mean,
                                                      invstd, gamma, beta,
                                                      (__bf16*)y, n, h,
                                                      relu ? 1 : 0);
  }
}

void launch_bn_bwd_partials_only16(const void* g, const void* x,
                                   const void* y, const float* mean,
                                   const float* invstd, long n, int h,
                                   bool relu, float* partials, float keep_inv,
                                   hipStream_t s) {
  if (n == 0) {
    HIP_CHECK(hipMemsetAsync(partials, 0, 2 * h * sizeof(float), s));
    return;
  }
  bn_bwd_partials_dispatch((const __bf16*)g, (const __bf16*)x,
                           (const __bf16*)y, mean, invstd, n, h, relu,
                           partials, keep_inv, s);
}

void launch_bn_bwd_apply_only16(const void* g, const void* x, const void* y,
                                const float* mean, const float* invstd,
                                const float* gamma, const float* partials,
                                const float* count_ptr, void* dx, long n,
                                int h, bool relu, float keep_inv,
                                hipStream_t s) {
  if (n == 0) return;
  bn_bwd_apply_kernel<<<grid_for(n * h), 256, 0, s>>>(
      (const __bf16*)g, (const __bf16*)x, (const __bf16*)y, mean, invstd,
      gamma, partials, (__bf16*)dx, n, n, h, relu ? 1 : 0, count_ptr,
      keep_inv);
}

void launch_bn_bwd16(const void* g, const void* x, const void* y,
                     const float* mean, const float* invstd,
                     const float* gamma, float* partials, void* dx,
                     float* dgamma, float* dbeta, long n, int h, bool relu,
                     float keep_inv, hipStream_t s) {
  if (n == 0) return;
  bn_bwd_partials_dispatch((const __bf16*)g, (const __bf16*)x,
                           (const __bf16*)y, mean, invstd, n, h, relu,
                           partials, keep_inv, s);
  bn_bwd_apply_kernel<<<grid_for(n * h), 256, 0, s>>>(
      (const __bf16*)g, (const __bf16*)x, (const __bf16*)y, mean, invstd,
      gamma, partials, (__bf16*)dx, n, n, h, relu ? 1 : 0, nullptr,
      keep_inv);
  bn_grad_affine_kernel<<<ceil_div(h, 256), 256, 0, s>>>(partials, dgamma,
                                                         dbeta, h);
}

// ---------------------------------------------------------------------------
// quantile loss + eval metrics
// ---------------------------------------------------------------------------

__global__ void quantile_loss_fwd_kernel(const float* __restrict__ y,
                                         const float* __restrict__ y_hat,
                                         float* __restrict__ out, long b,
                                         float tau) {
  __shared__ float smem[256];
  float acc = 0.f;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < b;
       i += (long)gridDim.x * blockDim.x) {
    const float e = y[i] - y_hat[i];
    acc += fmaxf(tau * e, (tau - 1.f) * e);
  }
  smem[threadIdx.x] = acc;
  __syncthreads();
  for (int off = blockDim.x / 2; off > 0; off >>= 1) {
    if (threadIdx.x < off) smem[threadIdx.x] += smem[threadIdx.x + off];
    __syncthreads();
  }
  if (threadIdx.x == 0) atomicAdd(out, smem[0] / b);
}

__global__ void quantile_loss_bwd_kernel(const float* __restrict__ g,
                                         const float* __restrict__ y,
                                         const float* __restrict__ y_hat,
                                         float* __restrict__ dy_hat, long b,
                                         float tau) {
  const float gs = g[0] / b;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < b;
       i += (long)gridDim.x * blockDim.x) {
    const float e = y[i] - y_hat[i];
    float d;
    if (e > 0.f) d = -tau;
    else if (e < 0.f) d = 1.f - tau;
    else d = 0.5f * (-tau) + 0.5f * (1.f - tau);  // torch.maximum tie split
    dy_hat[i] = gs * d;
  }
}

__global__ void eval_metrics_kernel(const float* __restrict__ y,
                                    const float* __restrict__ y_hat,
                                    float* __restrict__ out3, long b,
                                    float tau) {
  __shared__ float smem[3][256];
  float mae = 0.f, mape = 0.f, q = 0.f;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < b;
       i += (long)gridDim.x * blockDim.x) {
    const float err = y_hat[i] - y[i];
    const float a = fabsf(err);
    mae += a;
    mape += a / y[i];
    const float e = -err;
    q += fmaxf(tau * e, (tau - 1.f) * e);
  }
  smem[0][threadIdx.x] = mae;
  smem[1][threadIdx.x] = mape;
  smem[2][threadIdx.x] = q;
  __syncthreads();
  for (int off = blockDim.x / 2; off > 0; off >>= 1) {
    if (threadIdx.x < off)
      for (int k = 0; k < 3; ++k)
        smem[k][threadIdx.x] += smem[k][threadIdx.x + off];
    __syncthreads();
  }
  if (threadIdx.x == 0)
    for (int k = 0; k < 3; ++k) atomicAdd(&out3[k], smem[k][0]);
}

void launch_quantile_loss_fwd(const float* y, const float* y_hat, float* out,
                              long b, float tau, hipStream_t s) {
  HIP_CHECK(hipMemsetAsync(out, 0, sizeof(float), s));
  quantile_loss_fwd_kernel<<<grid_for(b, 256, 256), 256, 0, s>>>(y, y_hat, out,
                                                                 b, tau);
}
void launch_quantile_loss_bwd(const float* g, const float* y,
                              const float* y_hat, float* dy_hat, long b,
                              float tau, hipStream_t s) {
  quantile_loss_bwd_kernel<<<grid_for(b), 256, 0, s>>>(g, y, y_hat, dy_hat, b,
                                                       tau);
}
void launch_eval_metrics(const float* y, const float* y_hat, float* out3,
                         long b, float tau, hipStream_t s) {
  HIP_CHECK(hipMemsetAsync(out3, 0, 3 * sizeof(float), s));
  eval_metrics_kernel<<<grid_for(b, 256, 256), 256, 0, s>>>(y, y_hat, out3, b,
                                                            tau);
}

// ---------------------------------------------------------------------------
// Adam on flat master buffers (torch.optim.Adam math, K14)
// ---------------------------------------------------------------------------

// step/bias state lives ON DEVICE (state[0]=step, state[1]=1-b1^t,
// state[2]=1-b2^t) so the whole Adam update is hipGraph-replayable: each
// replay increments the step and recomputes the bias corrections.
// sstate (dynamic loss scaling, all device-resident so the whole step is
// hipGraph-replayable with the scale evolving across replays):
//   sstate[0] = current scale, sstate[1] = clean-step counter,
//   sstate[2] = found_inf flag for THIS step.
// A skipped step (non-finite grads) advances neither the Adam moments nor
// the bias-correction counter — torch.cuda.amp.GradScaler semantics.
__global__ void adam_tick_kernel(float* __restrict__ state, float b1,
                                 float b2,
                                 const float* __restrict__ sstate = nullptr) {
  if (sstate && sstate[2] != 0.f) return;  // skipped step
  const float t = state[0] + 1.f;
  state[0] = t;
  state[1] = 1.f - powf(b1, t);
  state[2] = 1.f - powf(b2, t);
}

__global__ void adam_kernel(float* __restrict__ p, const float* __restrict__ g,
                            float* __restrict__ m, float* __restrict__ v,
                            const float* __restrict__ state, long numel,
                            float lr, float b1, float b2, float eps,
                            float gscale,
                            const float* __restrict__ sstate = nullptr) {
  if (sstate && sstate[2] != 0.f) return;  // skipped step
  const long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  const float step_size = lr / state[1];
  const float sqrt_bias2 = sqrtf(state[2]);
  const float gs = sstate ? 1.f / sstate[0] : gscale;
  for (long t = i0; t < numel; t += stride) {
    const float gt = g[t] * gs;  // loss-scaling unscale
    const float mt = b1 * m[t] + (1.f - b1) * gt;
    const float vt = b2 * v[t] + (1.f - b2) * gt * gt;
    m[t] = mt;
    v[t] = vt;
    const float denom = sqrtf(vt) / sqrt_bias2 + eps;
    p[t] -= step_size * mt / denom;
  }
}

__global__ void scaler_begin_kernel(float* __restrict__ sstate) {
  sstate[2] = 0.f;
}

__global__ void grad_scan_kernel(const float* __restrict__ g, long numel,
                                 float* __restrict__ sstate) {
  const long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  float bad = 0.f;
  for (long t = i0; t < numel; t += stride) {
    const float x = g[t];
    if (!isfinite(x)) { bad = 1.f; break; }
  }
  if (bad != 0.f) sstate[2] = 1.f;  // benign race: every writer stores 1
}

__global__ void scaler_update_kernel(float* __restrict__ sstate, float backoff,
                                     float growth, float interval) {
  if (sstate[2] != 0.f) {
    sstate[0] = fmaxf(sstate[0] * backoff, 1.f);
    sstate[1] = 0.f;
  } else {
    sstate[1] += 1.f;
    if (sstate[1] >= interval) {
      sstate[0] = fminf(sstate[0] * growth, 4294967296.f);
      sstate[1] = 0.f;
    }
  }
}

void launch_adam(float* p, const float* g, float* m, float* v, float* state,
                 long numel, float lr, float b1, float b2, float eps, float gscale,
                 hipStream_t s) {
  if (numel == 0) return;
  adam_tick_kernel<<<1, 1, 0, s>>>(state, b1, b2, nullptr);
  adam_kernel<<<grid_for(numel), 256, 0, s>>>(p, g, m, v, state, numel, lr,
                                              b1, b2, eps, gscale, nullptr);
}

void launch_adam_dynamic(float* p, const float* g, float* m, float* v,
                         float* state, float* sstate, long numel, float lr,
                         float b1, float b2, float eps, float backoff,
                         float growth, float interval, hipStream_t s) {
  if (numel == 0) return;
  scaler_begin_kernel<<<1, 1, 0, s>>>(sstate);
  grad_scan_kernel<<<grid_for(numel), 256, 0, s>>>(g, numel, sstate);
  adam_tick_kernel<<<1, 1, 0, s>>>(state, b1, b2, sstate);
  adam_kernel<<<grid_for(numel), 256, 0, s>>>(p, g, m, v, state, numel, lr,
                                              b1, b2, eps, 1.f, sstate);
  scaler_update_kernel<<<1, 1, 0, s>>>(sstate, backoff, growth, interval);
}
