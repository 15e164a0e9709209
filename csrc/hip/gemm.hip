// Hand-written MFMA GEMM suite for the dense node-feature linears (gfx950).
//
// The model's GEMMs (reference K2) are skinny: M = nodes/edges
// (1e4..1e5), N/K in {9+H, H, 2H, 4H} with H up to 512.  Exact-fp32 path uses
// v_mfma_f32_16x16x4_f32 (the f32 "SGEMM" MFMA — 157 TF peak, bitwise equal
// to an fmaf chain, guide §3), so numerics match the eager oracle to fp32
// roundoff.  Three layouts cover forward + backward:
//
//   NT: C[M,N] = A[M,K] @ B[N,K]^T   (x @ W^T — forward; W torch layout)
//   NN: C[M,K] = A[M,N] @ B[N,K]     (g @ W — dgrad)
//   TN: C[N,K] = A[M,N]^T @ B[M,K]   (g^T @ x — wgrad, contraction over M,
//                                     split-K + fused bias-grad column sums)
//
// Tiles are templated: 64x64 (4 waves of 32x32) for small shapes, 128x128
// (4 waves of 64x64, 64 acc VGPRs) for the big node GEMMs.  BK=32; LDS is
// staged [contract][free] so every MFMA fragment read is conflict-free
// (16 lanes read 16 consecutive floats of one LDS row).

#include "common.h"
#include <cstdlib>

// PERTGNN_DETERMINISTIC=1: collapse split-K wgrad to one slice so the dw/db
// reduction order is fixed (read per call — tests flip it at runtime)
static inline bool pertgnn_deterministic() {
  const char* e = getenv("PERTGNN_DETERMINISTIC");
  return e && e[0] == '1';
}

#define GEMM_BK 32
#define GEMM_THREADS 256

typedef __attribute__((ext_vector_type(4))) float f32x4;

// ---------------------------------------------------------------------------
// LDS staging: 256 threads stage a [BF x BK] (free x contract) global tile
// into lds[BK][BF].  BF in {64, 128}.
// ---------------------------------------------------------------------------

// operand stored [free][contract] in global: transpose-stage.
template <int BF>
__device__ __forceinline__ void stage_transpose(
    const float* __restrict__ g, long ld, int free0, int contract0,
    int free_max, int contract_max, float* lds) {
  const int t = threadIdx.x;
  const int fr = t / 8;           // 32 free rows per pass
  const int cq = (t % 8) * 4;     // contract quad
  const bool interior = (free0 + BF <= free_max) &&
                        (contract0 + GEMM_BK <= contract_max);
  if (interior && (ld & 3) == 0) {
#pragma unroll
    for (int half = 0; half < BF / 32; ++half) {
      const int f = fr + half * 32;
      const f32x4 v = *reinterpret_cast<const f32x4*>(
          &g[(long)(free0 + f) * ld + contract0 + cq]);
#pragma unroll
      for (int u = 0; u < 4; ++u) lds[(cq + u) * BF + f] = v[u];
    }
    return;
  }
  if (interior) {  // odd leading dim: unchecked scalar loads
#pragma unroll
    for (int half = 0; half < BF / 32; ++half) {
      const int f = fr + half * 32;
      const float* row = &g[(long)(free0 + f) * ld + contract0 + cq];
#pragma unroll
      for (int u = 0; u < 4; ++u) lds[(cq + u) * BF + f] = row[u];
    }
    return;
  }
#pragma unroll
  for (int half = 0; half < BF / 32; ++half) {
    const int f = fr + half * 32;
    const int gf = free0 + f;
    float v[4];
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      const int gc = contract0 + cq + u;
      v[u] = (gf < free_max && gc < contract_max) ? g[(long)gf * ld + gc] : 0.f;
    }
#pragma unroll
    for (int u = 0; u < 4; ++u) lds[(cq + u) * BF + f] = v[u];
  }
}

// operand stored [contract][free] in global: direct stage.
template <int BF>
__device__ __forceinline__ void stage_direct(
    const float* __restrict__ g, long ld, int contract0, int free0,
    int contract_max, int free_max, float* lds) {
  const int t = threadIdx.x;
  constexpr int QUADS = BF / 4;           // float4 slots per contract row
  const int c = t / QUADS;                // contract rows per pass
  const int fq = (t % QUADS) * 4;
  constexpr int CSTEP = GEMM_THREADS / QUADS;
  const bool interior = (contract0 + GEMM_BK <= contract_max) &&
                        (free0 + BF <= free_max);
  if (interior && ((ld & 3) == 0) && ((free0 & 3) == 0)) {
#pragma unroll
    for (int half = 0; half < GEMM_BK / CSTEP; ++half) {
      const int cc = c + half * CSTEP;
      const f32x4 v = *reinterpret_cast<const f32x4*>(
          &g[(long)(contract0 + cc) * ld + free0 + fq]);
      *reinterpret_cast<f32x4*>(&lds[cc * BF + fq]) = v;
    }
    return;
  }
  if (interior) {
#pragma unroll
    for (int half = 0; half < GEMM_BK / CSTEP; ++half) {
      const int cc = c + half * CSTEP;
      const float* row = &g[(long)(contract0 + cc) * ld + free0 + fq];
#pragma unroll
      for (int u = 0; u < 4; ++u) lds[cc * BF + fq + u] = row[u];
    }
    return;
  }
#pragma unroll
  for (int half = 0; half < GEMM_BK / CSTEP; ++half) {
    const int cc = c + half * CSTEP;
    const int gc = contract0 + cc;
    float v[4];
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      const int gf = free0 + fq + u;
      v[u] = (gc < contract_max && gf < free_max) ? g[(long)gc * ld + gf] : 0.f;
    }
#pragma unroll
    for (int u = 0; u < 4; ++u) lds[cc * BF + fq + u] = v[u];
  }
}

// ---------------------------------------------------------------------------
// wave tile: FM x FN fragments of 16x16 (wave covers 16FM x 16FN outputs)
// ---------------------------------------------------------------------------

template <int FM, int FN, int BM, int BN>
struct WaveTileT {
  f32x4 acc[FM][FN];
  __device__ __forceinline__ void zero() {
#pragma unroll
    for (int i = 0; i < FM; ++i)
#pragma unroll
      for (int j = 0; j < FN; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};
  }
  __device__ __forceinline__ void mma(const float* lds_a, const float* lds_b,
                                      int wm, int wn, int lane) {
    const int fi = lane & 15;
    const int fk = lane >> 4;
#pragma unroll
    for (int s = 0; s < GEMM_BK / 4; ++s) {
      const int k = s * 4 + fk;
      float a[FM], b[FN];
#pragma unroll
      for (int mi = 0; mi < FM; ++mi) a[mi] = lds_a[k * BM + wm + mi * 16 + fi];
#pragma unroll
      for (int ni = 0; ni < FN; ++ni) b[ni] = lds_b[k * BN + wn + ni * 16 + fi];
#pragma unroll
      for (int mi = 0; mi < FM; ++mi)
#pragma unroll
        for (int ni = 0; ni < FN; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x4f32(a[mi], b[ni],
                                                             acc[mi][ni], 0,
                                                             0, 0);
    }
  }
  __device__ __forceinline__ void store(float* __restrict__ c, long ldc,
                                        int row0, int col0, int m_max,
                                        int n_max, const float* bias, int relu,
                                        int lane) {
    const int fcol = lane & 15;
    const int frow = (lane >> 4) * 4;
#pragma unroll
    for (int mi = 0; mi < FM; ++mi)
#pragma unroll
      for (int ni = 0; ni < FN; ++ni)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int row = row0 + mi * 16 + frow + r;
          const int col = col0 + ni * 16 + fcol;
          if (row < m_max && col < n_max) {
            float v = acc[mi][ni][r];
            if (bias) v += bias[col];
            if (relu) v = fmaxf(v, 0.f);
            c[(long)row * ldc + col] = v;
          }
        }
  }
};

// ---------------------------------------------------------------------------
// NT kernel (both operands [free][contract] in global), templated tile
// ---------------------------------------------------------------------------

template <int BM, int BN>
__launch_bounds__(GEMM_THREADS)
__global__ void gemm_f32_nt_kernel(const float* __restrict__ a,
                                   const float* __restrict__ b,
                                   const float* __restrict__ bias,
                                   float* __restrict__ c, int m, int n, int k,
                                   int relu) {
  constexpr int FM = BM / 32, FN = BN / 32;  // 2x2 wave grid
  __shared__ float lds_a[2][GEMM_BK * BM];
  __shared__ float lds_b[2][GEMM_BK * BN];
  const int bid = xcd_swizzle(blockIdx.x, gridDim.x);
  const int tiles_n = (n + BN - 1) / BN;
  const int m0 = (bid / tiles_n) * BM;
  const int n0 = (bid % tiles_n) * BN;
  const int wave = threadIdx.x / PERTGNN_WAVE;
  const int lane = threadIdx.x % PERTGNN_WAVE;
  const int wm = (wave >> 1) * (BM / 2);
  const int wn = (wave & 1) * (BN / 2);

  WaveTileT<FM, FN, BM, BN> wt;
  wt.zero();
  int buf = 0;
  stage_transpose<BM>(a, k, m0, 0, m, k, lds_a[0]);
  stage_transpose<BN>(b, k, n0, 0, n, k, lds_b[0]);
  __syncthreads();
  for (int k0 = GEMM_BK; k0 < k; k0 += GEMM_BK) {
    stage_transpose<BM>(a, k, m0, k0, m, k, lds_a[buf ^ 1]);
    stage_transpose<BN>(b, k, n0, k0, n, k, lds_b[buf ^ 1]);
    wt.mma(lds_a[buf], lds_b[buf], wm, wn, lane);
    __syncthreads();
    buf ^= 1;
  }
  wt.mma(lds_a[buf], lds_b[buf], wm, wn, lane);
  wt.store(c, n, m0 + wm, n0 + wn, m, n, bias, relu, lane);
}

// NN kernel: A [free][contract] (transpose-stage), B [contract][free] (direct)
template <int BM, int BN>
__launch_bounds__(GEMM_THREADS)
__global__ void gemm_f32_nn_kernel(const float* __restrict__ a,
                                   const float* __restrict__ b,
                                   const float* __restrict__ bias,
                                   float* __restrict__ c, int m, int n, int k2,
                                   int relu) {
  constexpr int FM = BM / 32, FN = BN / 32;
  __shared__ float lds_a[2][GEMM_BK * BM];
  __shared__ float lds_b[2][GEMM_BK * BN];
  const int bid = xcd_swizzle(blockIdx.x, gridDim.x);
  const int tiles_n = (k2 + BN - 1) / BN;
  const int m0 = (bid / tiles_n) * BM;
  const int n0 = (bid % tiles_n) * BN;
  const int wave = threadIdx.x / PERTGNN_WAVE;
  const int lane = threadIdx.x % PERTGNN_WAVE;
  const int wm = (wave >> 1) * (BM / 2);
  const int wn = (wave & 1) * (BN / 2);

  WaveTileT<FM, FN, BM, BN> wt;
  wt.zero();
  int buf = 0;
  stage_transpose<BM>(a, n, m0, 0, m, n, lds_a[0]);
  stage_direct<BN>(b, k2, 0, n0, n, k2, lds_b[0]);
  __syncthreads();
  for (int c0 = GEMM_BK; c0 < n; c0 += GEMM_BK) {
    stage_transpose<BM>(a, n, m0, c0, m, n, lds_a[buf ^ 1]);
    stage_direct<BN>(b, k2, c0, n0, n, k2, lds_b[buf ^ 1]);
    wt.mma(lds_a[buf], lds_b[buf], wm, wn, lane);
    __syncthreads();
    buf ^= 1;
  }
  wt.mma(lds_a[buf], lds_b[buf], wm, wn, lane);
  wt.store(c, k2, m0 + wm, n0 + wn, m, k2, bias, relu, lane);
}

// TN kernel with split-K over M; fused dbias column sums (tile_k == 0 blocks).
template <int BM, int BN>
__launch_bounds__(GEMM_THREADS)
__global__ void gemm_f32_tn_kernel(const float* __restrict__ a,
                                   const float* __restrict__ b,
                                   float* __restrict__ c,
                                   float* __restrict__ dbias, int m, int n,
                                   int k2, int slices) {
  constexpr int FM = BM / 32, FN = BN / 32;
  __shared__ float lds_a[2][GEMM_BK * BM];
  __shared__ float lds_b[2][GEMM_BK * BN];
  const int bid = xcd_swizzle(blockIdx.x, gridDim.x);
  const int tiles_k = (k2 + BN - 1) / BN;
  const int tile_id = bid / slices;
  const int slice = bid % slices;
  const int n0 = (tile_id / tiles_k) * BM;
  const int k0 = (tile_id % tiles_k) * BN;
  if (n0 >= n) return;
  const int wave = threadIdx.x / PERTGNN_WAVE;
  const int lane = threadIdx.x % PERTGNN_WAVE;
  const int wm = (wave >> 1) * (BM / 2);
  const int wn = (wave & 1) * (BN / 2);

  const int per_slice =
      ((m + slices - 1) / slices + GEMM_BK - 1) / GEMM_BK * GEMM_BK;
  const int c_beg = slice * per_slice;
  const int c_end = min(m, c_beg + per_slice);
  if (c_beg >= c_end) return;

  WaveTileT<FM, FN, BM, BN> wt;
  wt.zero();
  const bool do_bias = (dbias != nullptr) && (k0 == 0);
  float dbsum = 0.f;
  const int bcol = threadIdx.x;
  int buf = 0;
  stage_direct<BM>(a, n, c_beg, n0, c_end, n, lds_a[0]);
  stage_direct<BN>(b, k2, c_beg, k0, c_end, k2, lds_b[0]);
  __syncthreads();
  for (int cc = c_beg + GEMM_BK; cc < c_end; cc += GEMM_BK) {
    stage_direct<BM>(a, n, cc, n0, c_end, n, lds_a[buf ^ 1]);
    stage_direct<BN>(b, k2, cc, k0, c_end, k2, lds_b[buf ^ 1]);
    wt.mma(lds_a[buf], lds_b[buf], wm, wn, lane);
    if (do_bias && bcol < BM)
#pragma unroll
      for (int r = 0; r < GEMM_BK; ++r) dbsum += lds_a[buf][r * BM + bcol];
    __syncthreads();
    buf ^= 1;
  }
  wt.mma(lds_a[buf], lds_b[buf], wm, wn, lane);
  if (do_bias && bcol < BM) {
#pragma unroll
    for (int r = 0; r < GEMM_BK; ++r) dbsum += lds_a[buf][r * BM + bcol];
    if (n0 + bcol < n) atomicAdd(&dbias[n0 + bcol], dbsum);
  }

  const int fcol = lane & 15;
  const int frow = (lane >> 4) * 4;
#pragma unroll
  for (int mi = 0; mi < FM; ++mi)
#pragma unroll
    for (int ni = 0; ni < FN; ++ni)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = n0 + wm + mi * 16 + frow + r;
        const int col = k0 + wn + ni * 16 + fcol;
        if (row < n && col < k2) {
          if (slices == 1)
            c[(long)row * k2 + col] = wt.acc[mi][ni][r];
          else
            atomicAdd(&c[(long)row * k2 + col], wt.acc[mi][ni][r]);
        }
      }
}

// column sum for standalone bias gradients
__global__ void colsum_par_kernel(const float* __restrict__ g,
                                  float* __restrict__ out, long m, int n) {
  extern __shared__ float smem[];
  for (int c = threadIdx.x; c < n; c += blockDim.x) smem[c] = 0.f;
  __syncthreads();
  const long rows_per_block = (m + gridDim.x - 1) / gridDim.x;
  const long r0 = (long)blockIdx.x * rows_per_block;
  const long r1 = min(m, r0 + rows_per_block);
  for (long r = r0; r < r1; ++r)
    for (int c = threadIdx.x; c < n; c += blockDim.x) smem[c] += g[r * n + c];
  __syncthreads();
  for (int c = threadIdx.x; c < n; c += blockDim.x)
    if (smem[c] != 0.f) atomicAdd(&out[c], smem[c]);
}

// ---------------------------------------------------------------------------
// launchers — pick the 128 tile for big shapes, 64 otherwise
// ---------------------------------------------------------------------------

void launch_gemm_f32_nt(const float* a, const float* b, const float* bias,
                        float* c, int m, int n, int k, bool relu,
                        hipStream_t s) {
  if (m >= 512 && n >= 128) {
    const int grid = ((m + 127) / 128) * ((n + 127) / 128);
    gemm_f32_nt_kernel<128, 128><<<dim3(grid), dim3(GEMM_THREADS), 0, s>>>(
        a, b, bias, c, m, n, k, relu ? 1 : 0);
  } else {
    const int grid = ((m + 63) / 64) * ((n + 63) / 64);
    gemm_f32_nt_kernel<64, 64><<<dim3(grid), dim3(GEMM_THREADS), 0, s>>>(
        a, b, bias, c, m, n, k, relu ? 1 : 0);
  }
}

void launch_gemm_f32_nn(const float* a, const float* b, const float* bias,
                        float* c, int m, int n, int k2, bool relu,
                        hipStream_t s) {
  if (m >= 512 && k2 >= 128) {
    const int grid = ((m + 127) / 128) * ((k2 + 127) / 128);
    gemm_f32_nn_kernel<128, 128><<<dim3(grid), dim3(GEMM_THREADS), 0, s>>>(
        a, b, bias, c, m, n, k2, relu ? 1 : 0);
  } else {
    const int grid = ((m + 63) / 64) * ((k2 + 63) / 64);
    gemm_f32_nn_kernel<64, 64><<<dim3(grid), dim3(GEMM_THREADS), 0, s>>>(
        a, b, bias, c, m, n, k2, relu ? 1 : 0);
  }
}

void launch_gemm_f32_tn(const float* a, const float* b, float* c, float* dbias,
                        int m, int n, int k2, hipStream_t s) {
  const bool big = (n >= 128 && k2 >= 128);
  const int bm = big ? 128 : 64;
  const int bn = big ? 128 : 64;
  const int tiles = ((n + bm - 1) / bm) * ((k2 + bn - 1) / bn);
  int slices = 1;
  while (!pertgnn_deterministic() && tiles * slices < 512 && slices < 64 &&
         (long)slices * GEMM_BK * 4 < m)
    slices *= 2;
  if (slices > 1)
    HIP_CHECK(hipMemsetAsync(c, 0, (long)n * k2 * sizeof(float), s));
  if (dbias) HIP_CHECK(hipMemsetAsync(dbias, 0, n * sizeof(float), s));
  if (big)
    gemm_f32_tn_kernel<128, 128>
        <<<dim3(tiles * slices), dim3(GEMM_THREADS), 0, s>>>(a, b, c, dbias, m,
                                                             n, k2, slices);
  else
    gemm_f32_tn_kernel<64, 64>
        <<<dim3(tiles * slices), dim3(GEMM_THREADS), 0, s>>>(a, b, c, dbias, m,
                                                             n, k2, slices);
}

void launch_colsum(const float* g, float* out, long m, int n, hipStream_t s) {
  HIP_CHECK(hipMemsetAsync(out, 0, n * sizeof(float), s));
  const int blocks = (int)min((long)256, (m + 63) / 64);
  colsum_par_kernel<<<dim3(blocks), dim3(256), n * sizeof(float), s>>>(g, out,
                                                                       m, n);
}
