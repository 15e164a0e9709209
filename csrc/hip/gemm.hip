// Hand-written MFMA GEMM suite for the dense node-feature linears (gfx950).
//
// The model's GEMMs (reference K2) are skinny: M = nodes/edges
// (1e4..1e5), K,N in {9+H, H, 2H, 1} with H up to 512.  Exact-fp32 path uses
// v_mfma_f32_16x16x4_f32 (the f32 "SGEMM" MFMA — 157 TF peak, bitwise equal
// to an fmaf chain, guide §3), so numerics match the eager oracle to fp32
// roundoff.  Three layouts cover forward + backward:
//
//   NT: C[M,N] = A[M,K] @ B[N,K]^T   (x @ W^T — forward; W torch layout)
//   NN: C[M,K] = A[M,N] @ B[N,K]     (g @ W — dgrad)
//   TN: C[N,K] = A[M,N]^T @ B[M,K]   (g^T @ x — wgrad, contraction over M)
//
// Geometry: 64x64 block tile, BK=32, 4 waves (2x2 of 32x32 wave tiles), LDS
// staged [contract][free] so every MFMA fragment read is bank-conflict-free
// (16 lanes read 16 consecutive floats of one LDS row).
//
// Epilogues: +bias[N], optional ReLU (fuses reference K8 into K2).

#include "common.h"

#define GEMM_BM 64
#define GEMM_BN 64
#define GEMM_BK 32
#define GEMM_THREADS 256

typedef __attribute__((ext_vector_type(4))) float f32x4;

// ---------------------------------------------------------------------------
// LDS staging helpers. 256 threads stage a 64x32 (free x contract) global
// tile into lds[contract][free] ([32][64] f32 = 8 KiB).
// ---------------------------------------------------------------------------

// operand stored [free][contract] in global (row stride = ld): transpose-stage.
// thread t loads a float4 along contract; zero-fill outside bounds.  Interior
// tiles with 16B-alignable rows (ld % 4 == 0) take the vectorized fast path.
__device__ __forceinline__ void stage_transpose(
    const float* __restrict__ g, long ld, int free0, int contract0,
    int free_max, int contract_max, float* lds /* [GEMM_BK][GEMM_BM] */) {
  const int t = threadIdx.x;
  const int fr = t / 8;           // 32 rows per pass; two passes cover 64
  const int cq = (t % 8) * 4;     // contract quad
  const bool fast = (free0 + GEMM_BM <= free_max) &&
                    (contract0 + GEMM_BK <= contract_max) && ((ld & 3) == 0);
  if (fast) {
#pragma unroll
    for (int half = 0; half < 2; ++half) {
      const int f = fr + half * 32;
      const f32x4 v = *reinterpret_cast<const f32x4*>(
          &g[(long)(free0 + f) * ld + contract0 + cq]);
#pragma unroll
      for (int u = 0; u < 4; ++u) lds[(cq + u) * GEMM_BM + f] = v[u];
    }
    return;
  }
#pragma unroll
  for (int half = 0; half < 2; ++half) {
    const int f = fr + half * 32;
    const int gf = free0 + f;
    float v[4];
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      const int gc = contract0 + cq + u;
      v[u] = (gf < free_max && gc < contract_max) ? g[(long)gf * ld + gc] : 0.f;
    }
#pragma unroll
    for (int u = 0; u < 4; ++u) lds[(cq + u) * GEMM_BM + f] = v[u];
  }
}

// operand stored [contract][free] in global (row stride = ld): direct stage.
__device__ __forceinline__ void stage_direct(
    const float* __restrict__ g, long ld, int contract0, int free0,
    int contract_max, int free_max, float* lds /* [GEMM_BK][GEMM_BM] */) {
  const int t = threadIdx.x;
  const int c = t / 16;            // two passes cover 32 contract rows
  const int fq = (t % 16) * 4;     // free quad
  const bool fast = (contract0 + GEMM_BK <= contract_max) &&
                    (free0 + GEMM_BM <= free_max) && ((ld & 3) == 0) &&
                    ((free0 & 3) == 0);
  if (fast) {
#pragma unroll
    for (int half = 0; half < 2; ++half) {
      const int cc = c + half * 16;
      const f32x4 v = *reinterpret_cast<const f32x4*>(
          &g[(long)(contract0 + cc) * ld + free0 + fq]);
      *reinterpret_cast<f32x4*>(&lds[cc * GEMM_BM + fq]) = v;
    }
    return;
  }
#pragma unroll
  for (int half = 0; half < 2; ++half) {
    const int cc = c + half * 16;
    const int gc = contract0 + cc;
    float v[4];
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      const int gf = free0 + fq + u;
      v[u] = (gc < contract_max && gf < free_max) ? g[(long)gc * ld + gf] : 0.f;
    }
#pragma unroll
    for (int u = 0; u < 4; ++u) lds[cc * GEMM_BM + fq + u] = v[u];
  }
}

// ---------------------------------------------------------------------------
// core MFMA tile compute: both operands in lds[GEMM_BK][64]
// ---------------------------------------------------------------------------

struct WaveTile {
  f32x4 acc[2][2];  // [mi][ni] 16x16 fragments of the 32x32 wave tile
  __device__ __forceinline__ void zero() {
#pragma unroll
    for (int i = 0; i < 2; ++i)
#pragma unroll
      for (int j = 0; j < 2; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};
  }
  __device__ __forceinline__ void mma(const float* lds_a, const float* lds_b,
                                      int wm, int wn, int lane) {
    const int fi = lane & 15;   // fragment row (A) / col (B,C)
    const int fk = lane >> 4;   // fragment k
#pragma unroll
    for (int s = 0; s < GEMM_BK / 4; ++s) {
      const int k = s * 4 + fk;
#pragma unroll
      for (int mi = 0; mi < 2; ++mi) {
        const float a = lds_a[k * GEMM_BM + wm + mi * 16 + fi];
#pragma unroll
        for (int ni = 0; ni < 2; ++ni) {
          const float b = lds_b[k * GEMM_BM + wn + ni * 16 + fi];
          acc[mi][ni] =
              __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc[mi][ni], 0, 0, 0);
        }
      }
    }
  }
  // write C tile with optional bias[col] and relu
  __device__ __forceinline__ void store(float* __restrict__ c, long ldc,
                                        int row0, int col0, int m_max,
                                        int n_max, const float* bias, int relu,
                                        int lane) {
    const int fcol = lane & 15;
    const int frow = (lane >> 4) * 4;
#pragma unroll
    for (int mi = 0; mi < 2; ++mi)
#pragma unroll
      for (int ni = 0; ni < 2; ++ni)
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
// NT: C[M,N] = A[M,K] @ B[N,K]^T  — both operands [free][contract] in global
// ---------------------------------------------------------------------------

__launch_bounds__(GEMM_THREADS)
__global__ void gemm_f32_nt_kernel(const float* __restrict__ a,
                                   const float* __restrict__ b,
                                   const float* __restrict__ bias,
                                   float* __restrict__ c, int m, int n, int k,
                                   int relu) {
  __shared__ float lds_a[2][GEMM_BK * GEMM_BM];
  __shared__ float lds_b[2][GEMM_BK * GEMM_BM];
  const int tiles_n = (n + GEMM_BN - 1) / GEMM_BN;
  const int tile_m = blockIdx.x / tiles_n;
  const int tile_n = blockIdx.x % tiles_n;
  const int m0 = tile_m * GEMM_BM;
  const int n0 = tile_n * GEMM_BN;
  const int wave = threadIdx.x / PERTGNN_WAVE;
  const int lane = threadIdx.x % PERTGNN_WAVE;
  const int wm = (wave >> 1) * 32;  // 2x2 wave grid
  const int wn = (wave & 1) * 32;

  WaveTile wt;
  wt.zero();
  int buf = 0;
  stage_transpose(a, k, m0, 0, m, k, lds_a[0]);
  stage_transpose(b, k, n0, 0, n, k, lds_b[0]);
  __syncthreads();
  for (int k0 = GEMM_BK; k0 < k; k0 += GEMM_BK) {
    stage_transpose(a, k, m0, k0, m, k, lds_a[buf ^ 1]);
    stage_transpose(b, k, n0, k0, n, k, lds_b[buf ^ 1]);
    wt.mma(lds_a[buf], lds_b[buf], wm, wn, lane);
    __syncthreads();
    buf ^= 1;
  }
  wt.mma(lds_a[buf], lds_b[buf], wm, wn, lane);
  wt.store(c, n, m0 + wm, n0 + wn, m, n, bias, relu, lane);
}

// ---------------------------------------------------------------------------
// NN: C[M,K2] = A[M,N] @ B[N,K2] — A is [free][contract], B is [contract][free]
// ---------------------------------------------------------------------------

__launch_bounds__(GEMM_THREADS)
__global__ void gemm_f32_nn_kernel(const float* __restrict__ a,
                                   const float* __restrict__ b,
                                   const float* __restrict__ bias,
                                   float* __restrict__ c, int m, int n, int k2,
                                   int relu) {
  __shared__ float lds_a[2][GEMM_BK * GEMM_BM];
  __shared__ float lds_b[2][GEMM_BK * GEMM_BM];
  const int tiles_n = (k2 + GEMM_BN - 1) / GEMM_BN;
  const int tile_m = blockIdx.x / tiles_n;
  const int tile_n = blockIdx.x % tiles_n;
  const int m0 = tile_m * GEMM_BM;
  const int n0 = tile_n * GEMM_BN;
  const int wave = threadIdx.x / PERTGNN_WAVE;
  const int lane = threadIdx.x % PERTGNN_WAVE;
  const int wm = (wave >> 1) * 32;
  const int wn = (wave & 1) * 32;

  WaveTile wt;
  wt.zero();
  int buf = 0;
  stage_transpose(a, n, m0, 0, m, n, lds_a[0]);
  stage_direct(b, k2, 0, n0, n, k2, lds_b[0]);
  __syncthreads();
  for (int c0 = GEMM_BK; c0 < n; c0 += GEMM_BK) {
    stage_transpose(a, n, m0, c0, m, n, lds_a[buf ^ 1]);
    stage_direct(b, k2, c0, n0, n, k2, lds_b[buf ^ 1]);
    wt.mma(lds_a[buf], lds_b[buf], wm, wn, lane);
    __syncthreads();
    buf ^= 1;
  }
  wt.mma(lds_a[buf], lds_b[buf], wm, wn, lane);
  wt.store(c, k2, m0 + wm, n0 + wn, m, k2, bias, relu, lane);
}

// ---------------------------------------------------------------------------
// TN with split-K over M: C[N,K2] = A[M,N]^T @ B[M,K2]
// contraction = M (huge), both operands [contract][free] in global.
// Each block owns an (n,k) tile and an M-slice; slices are reduced with
// fp32 atomicAdd into C (C zeroed first).  Non-deterministic only in the
// fp32 rounding order of K-slices (weight grads; acceptable — see tests).
// ---------------------------------------------------------------------------

__launch_bounds__(GEMM_THREADS)
__global__ void gemm_f32_tn_kernel(const float* __restrict__ a,
                                   const float* __restrict__ b,
                                   float* __restrict__ c,
                                   float* __restrict__ dbias, int m, int n,
                                   int k2, int slices) {
  __shared__ float lds_a[2][GEMM_BK * GEMM_BM];
  __shared__ float lds_b[2][GEMM_BK * GEMM_BM];
  const int tiles_k = (k2 + GEMM_BN - 1) / GEMM_BN;
  const int tiles_n = (n + GEMM_BM - 1) / GEMM_BM;
  const int tile_id = blockIdx.x / slices;
  const int slice = blockIdx.x % slices;
  const int tile_n = tile_id / tiles_k;
  const int tile_k = tile_id % tiles_k;
  if (tile_n >= tiles_n) return;
  const int n0 = tile_n * GEMM_BM;
  const int k0 = tile_k * GEMM_BN;
  const int wave = threadIdx.x / PERTGNN_WAVE;
  const int lane = threadIdx.x % PERTGNN_WAVE;
  const int wm = (wave >> 1) * 32;
  const int wn = (wave & 1) * 32;

  const int per_slice = ((m + slices - 1) / slices + GEMM_BK - 1) / GEMM_BK * GEMM_BK;
  const int c_beg = slice * per_slice;
  const int c_end = min(m, c_beg + per_slice);
  if (c_beg >= c_end) return;

  WaveTile wt;
  wt.zero();
  // bias grad rides along: tile_k==0 blocks column-sum their staged A (=g)
  // tiles straight from LDS (no extra global pass over g).
  const bool do_bias = (dbias != nullptr) && (tile_k == 0);
  float dbsum = 0.f;
  const int bcol = threadIdx.x;  // threads 0..63 own A-tile columns
  int buf = 0;
  stage_direct(a, n, c_beg, n0, c_end, n, lds_a[0]);
  stage_direct(b, k2, c_beg, k0, c_end, k2, lds_b[0]);
  __syncthreads();
  for (int cc = c_beg + GEMM_BK; cc < c_end; cc += GEMM_BK) {
    stage_direct(a, n, cc, n0, c_end, n, lds_a[buf ^ 1]);
    stage_direct(b, k2, cc, k0, c_end, k2, lds_b[buf ^ 1]);
    wt.mma(lds_a[buf], lds_b[buf], wm, wn, lane);
    if (do_bias && bcol < GEMM_BM)
#pragma unroll
      for (int r = 0; r < GEMM_BK; ++r) dbsum += lds_a[buf][r * GEMM_BM + bcol];
    __syncthreads();
    buf ^= 1;
  }
  wt.mma(lds_a[buf], lds_b[buf], wm, wn, lane);
  if (do_bias && bcol < GEMM_BM) {
#pragma unroll
    for (int r = 0; r < GEMM_BK; ++r) dbsum += lds_a[buf][r * GEMM_BM + bcol];
    if (n0 + bcol < n) atomicAdd(&dbias[n0 + bcol], dbsum);
  }

  // accumulate into C with atomics (one slice may be the only writer)
  const int fcol = lane & 15;
  const int frow = (lane >> 4) * 4;
#pragma unroll
  for (int mi = 0; mi < 2; ++mi)
#pragma unroll
    for (int ni = 0; ni < 2; ++ni)
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

// column sum for bias gradients: out[n] = sum_m g[m][n]
// blocks cover row slices, LDS partials, one atomic per channel per block
__global__ void colsum_par_kernel(const float* __restrict__ g,
                                  float* __restrict__ out, long m, int n) {
  extern __shared__ float smem[];  // [n]
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
// launchers
// ---------------------------------------------------------------------------

void launch_gemm_f32_nt(const float* a, const float* b, const float* bias,
                        float* c, int m, int n, int k, bool relu,
                        hipStream_t s) {
  const int tiles_m = (m + GEMM_BM - 1) / GEMM_BM;
  const int tiles_n = (n + GEMM_BN - 1) / GEMM_BN;
  gemm_f32_nt_kernel<<<dim3(tiles_m * tiles_n), dim3(GEMM_THREADS), 0, s>>>(
      a, b, bias, c, m, n, k, relu ? 1 : 0);
}

void launch_gemm_f32_nn(const float* a, const float* b, const float* bias,
                        float* c, int m, int n, int k2, bool relu,
                        hipStream_t s) {
  const int tiles_m = (m + GEMM_BM - 1) / GEMM_BM;
  const int tiles_n = (k2 + GEMM_BN - 1) / GEMM_BN;
  gemm_f32_nn_kernel<<<dim3(tiles_m * tiles_n), dim3(GEMM_THREADS), 0, s>>>(
      a, b, bias, c, m, n, k2, relu ? 1 : 0);
}

void launch_gemm_f32_tn(const float* a, const float* b, float* c, float* dbias,
                        int m, int n, int k2, hipStream_t s) {
  const int tiles_n = (n + GEMM_BM - 1) / GEMM_BM;
  const int tiles_k = (k2 + GEMM_BN - 1) / GEMM_BN;
  const int tiles = tiles_n * tiles_k;
  // pick slices so total blocks ~>= 2x CUs (512) for occupancy
  int slices = 1;
  while (tiles * slices < 512 && slices < 64 &&
         (long)slices * GEMM_BK * 4 < m) slices *= 2;
  if (slices > 1) HIP_CHECK(hipMemsetAsync(c, 0, (long)n * k2 * sizeof(float), s));
  if (dbias) HIP_CHECK(hipMemsetAsync(dbias, 0, n * sizeof(float), s));
  gemm_f32_tn_kernel<<<dim3(tiles * slices), dim3(GEMM_THREADS), 0, s>>>(
      a, b, c, dbias, m, n, k2, slices);
}

void launch_colsum(const float* g, float* out, long m, int n, hipStream_t s) {
  HIP_CHECK(hipMemsetAsync(out, 0, n * sizeof(float), s));
  const int blocks = (int)min((long)256, (m + 63) / 64);
  colsum_par_kernel<<<dim3(blocks), dim3(256), n * sizeof(float), s>>>(g, out,
                                                                       m, n);
}
