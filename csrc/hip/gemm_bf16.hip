// bf16-compute GEMM path (gfx950): fp32 operands in HBM, rounded to bf16
// during LDS staging, v_mfma_f32_16x16x32_bf16 matrix cores (≈2 PF ceiling,
// 16x the f32-MFMA rate), fp32 accumulate, fp32 output.  This is the
// mixed-precision mode of BASELINE configs 2/5: memory traffic identical to
// the f32 path (activations stay fp32 end-to-end), only the matmul operands
// are bf16-rounded.
//
// LDS layout: [free][BK+PAD] bf16 rows (k-contiguous), so each lane's MFMA
// fragment (8 consecutive k) is ONE ds_read_b128; the +8-element row pad
// makes the 16-lane fragment reads conflict-free (guide §6 G4).

#include "common.h"

// PERTGNN_DETERMINISTIC=1: collapse split-K wgrad to one slice so the dw/db
// reduction order is fixed (read per call — tests flip it at runtime)
static inline bool pertgnn_deterministic() {
  const char* e = getenv("PERTGNN_DETERMINISTIC");
  return e && e[0] == '1';
}
#include <cstdlib>

#define BGEMM_PAD 8
#define BGEMM_THREADS 256

typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;
typedef __attribute__((ext_vector_type(8))) _Float16 f16x8;
typedef __attribute__((ext_vector_type(4))) _Float16 f16x4;

// element-type traits: bf16 and fp16 share the whole kernel structure; only
// the packed vector types and the MFMA intrinsic differ.
template <typename T16> struct Vec16;
template <> struct Vec16<__bf16> {
  using v4 = bf16x4;
  using v8 = bf16x8;
  static __device__ __forceinline__ f32x4 mfma(v8 a, v8 b, f32x4 c) {
    return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
  }
};
template <> struct Vec16<_Float16> {
  using v4 = f16x4;
  using v8 = f16x8;
  static __device__ __forceinline__ f32x4 mfma(v8 a, v8 b, f32x4 c) {
    return __builtin_amdgcn_mfma_f32_16x16x32_f16(a, b, c, 0, 0, 0);
  }
};

template <typename T16>
__device__ __forceinline__ typename Vec16<T16>::v4 pack4(float a, float b,
                                                         float c, float d) {
  typename Vec16<T16>::v4 r;
  r[0] = (T16)a; r[1] = (T16)b; r[2] = (T16)c; r[3] = (T16)d;
  return r;
}

// 4 consecutive source elements -> float[4] (vector load for both dtypes)
__device__ __forceinline__ void ld4(const float* p, float (&v)[4]) {
  *reinterpret_cast<f32x4*>(v) = *reinterpret_cast<const f32x4*>(p);
}
__device__ __forceinline__ void ld4(const __bf16* p, float (&v)[4]) {
  const bf16x4 b = *reinterpret_cast<const bf16x4*>(p);
#pragma unroll
  for (int u = 0; u < 4; ++u) v[u] = (float)b[u];
}
__device__ __forceinline__ float ld1(const float* p) { return *p; }
__device__ __forceinline__ float ld1(const __bf16* p) { return (float)*p; }

// global [free][contract] (contract-minor) -> LDS [free][BK+PAD]
// 4 bf16 packed into one 8-byte LDS write (scalar u16 LDS writes are ~2x
// slower — guide G13 applies to LDS too).
//
// Split into load() / commit() so the mainloop can ISSUE the next tile's
// global loads, run the MFMA work on the current LDS buffer while they are
// in flight, and only then wait + write LDS: the fused form stalled every
// wave on s_waitcnt(vmcnt) BEFORE any MFMA issued (PMC: 72% SQ_WAIT_ANY on
// the NT kernel).
template <int BF, int BK, typename T16, typename TA,
          int THREADS = BGEMM_THREADS>
struct StageCmin {
  static constexpr int LDW = BK + BGEMM_PAD;
  static constexpr int QUADS = BK / 4;
  static constexpr int FSTEP = THREADS / QUADS;
  static constexpr int HALVES = BF / FSTEP;
  float v[HALVES][4];

  __device__ __forceinline__ void load(const TA* __restrict__ g, long ld,
                                       int free0, int contract0, int free_max,
                                       int contract_max) {
    const int t = threadIdx.x;
    const int f = t / QUADS;
    const int cq = (t % QUADS) * 4;
    const bool interior = (free0 + BF <= free_max) &&
                          (contract0 + BK <= contract_max);
    const bool aligned = (ld & 3) == 0;
    // path choice HOISTED outside the loops: a branch inside the staging
    // loop makes hipcc emit per-element load+vmcnt(0) chains
    // (guide §5 ".s-level traps" (c))
    if (interior && aligned) {
#pragma unroll
      for (int half = 0; half < HALVES; ++half)
        ld4(&g[(long)(free0 + f + half * FSTEP) * ld + contract0 + cq],
            v[half]);
    } else if (interior) {  // odd leading dim: unchecked scalars
#pragma unroll
      for (int half = 0; half < HALVES; ++half) {
        const TA* row =
            &g[(long)(free0 + f + half * FSTEP) * ld + contract0 + cq];
#pragma unroll
        for (int u = 0; u < 4; ++u) v[half][u] = ld1(row + u);
      }
    } else {
#pragma unroll
      for (int half = 0; half < HALVES; ++half) {
        const int gf = free0 + f + half * FSTEP;
#pragma unroll
        for (int u = 0; u < 4; ++u) {
          const int gc = contract0 + cq + u;
          v[half][u] = (gf < free_max && gc < contract_max)
                           ? ld1(&g[(long)gf * ld + gc])
                           : 0.f;
        }
      }
    }
  }

  __device__ __forceinline__ void commit(T16* lds) const {
    const int t = threadIdx.x;
    const int f = t / QUADS;
    const int cq = (t % QUADS) * 4;
#pragma unroll
    for (int half = 0; half < HALVES; ++half)
      *reinterpret_cast<typename Vec16<T16>::v4*>(
          &lds[(f + half * FSTEP) * LDW + cq]) =
          pack4<T16>(v[half][0], v[half][1], v[half][2], v[half][3]);
  }
};

template <int BF, int BK, typename T16, int THREADS = BGEMM_THREADS,
          typename TA>
__device__ __forceinline__ void bstage_cmin(const TA* __restrict__ g,
                                            long ld, int free0, int contract0,
                                            int free_max, int contract_max,
                                            T16* lds) {
  StageCmin<BF, BK, T16, TA, THREADS> s;
  s.load(g, ld, free0, contract0, free_max, contract_max);
  s.commit(lds);
}

// global [contract][free] (contract-major) -> LDS [free][BK+PAD]: each thread
// transposes a 4x4 block in registers (4 coalesced f32x4 loads from 4
// contract rows), then writes 4 packed 8-byte LDS rows.  Split load/commit
// for the same software-pipelining reason as StageCmin.
template <int BF, int BK, typename T16, typename TA,
          int THREADS = BGEMM_THREADS>
struct StageCmaj {
  static constexpr int LDW = BK + BGEMM_PAD;
  static constexpr int FQUADS = BF / 4;
  static constexpr int CSTEP = (THREADS / FQUADS) * 4;
  static constexpr int HALVES = (BK > CSTEP ? BK / CSTEP : 1);
  // with more threads than BK/4 * FQUADS lanes of work, the surplus threads
  // idle through staging (ACTIVE guarded below) but still hit the barrier
  static constexpr bool SURPLUS = CSTEP > BK;
  float v[HALVES][4][4];

  __device__ __forceinline__ void load(const TA* __restrict__ g, long ld,
                                       int contract0, int free0,
                                       int contract_max, int free_max) {
    const int t = threadIdx.x;
    const int cb = (t / FQUADS) * 4;
    const int fq = (t % FQUADS) * 4;
    if (SURPLUS && cb >= BK) return;
    const bool interior = (contract0 + BK <= contract_max) &&
                          (free0 + BF <= free_max);
    const bool aligned = ((ld & 3) == 0) && ((free0 & 3) == 0);
    // path choice hoisted outside the loops (guide §5 ".s-level traps" (c))
    if (interior && aligned) {
#pragma unroll
      for (int half = 0; half < HALVES; ++half)
#pragma unroll
        for (int u = 0; u < 4; ++u)
          ld4(&g[(long)(contract0 + cb + half * CSTEP + u) * ld + free0 + fq],
              v[half][u]);
    } else if (interior) {
#pragma unroll
      for (int half = 0; half < HALVES; ++half)
#pragma unroll
        for (int u = 0; u < 4; ++u) {
          const TA* row =
              &g[(long)(contract0 + cb + half * CSTEP + u) * ld + free0 + fq];
#pragma unroll
          for (int w = 0; w < 4; ++w) v[half][u][w] = ld1(row + w);
        }
    } else {
#pragma unroll
      for (int half = 0; half < HALVES; ++half) {
        const int cc = cb + half * CSTEP;
#pragma unroll
        for (int u = 0; u < 4; ++u) {
          const int gc = contract0 + cc + u;
#pragma unroll
          for (int w = 0; w < 4; ++w) {
            const int gf = free0 + fq + w;
            v[half][u][w] = (gc < contract_max && gf < free_max)
                                ? ld1(&g[(long)gc * ld + gf])
                                : 0.f;
          }
        }
      }
    }
  }

  __device__ __forceinline__ void commit(T16* lds) const {
    const int t = threadIdx.x;
    const int cb = (t / FQUADS) * 4;
    const int fq = (t % FQUADS) * 4;
    if (SURPLUS && cb >= BK) return;
#pragma unroll
    for (int half = 0; half < HALVES; ++half)
#pragma unroll
      for (int w = 0; w < 4; ++w)
        *reinterpret_cast<typename Vec16<T16>::v4*>(
            &lds[(fq + w) * LDW + cb + half * CSTEP]) =
            pack4<T16>(v[half][0][w], v[half][1][w], v[half][2][w],
                       v[half][3][w]);
  }
};

template <int BF, int BK, typename T16, int THREADS = BGEMM_THREADS,
          typename TA>
__device__ __forceinline__ void bstage_cmaj(const TA* __restrict__ g,
                                            long ld, int contract0, int free0,
                                            int contract_max, int free_max,
                                            T16* lds) {
  StageCmaj<BF, BK, T16, TA, THREADS> s;
  s.load(g, ld, contract0, free0, contract_max, free_max);
  s.commit(lds);
}

// ---------------------------------------------------------------------------
// wave tile: FM x FN fragments of 16x16; K-step 32 per MFMA, BK=64 = 2 steps
// fragment layout (mfma_f32_16x16x32_bf16): lane l holds 8 k-consecutive
// elements at k = (l>>4)*8 of row/col (l&15); C/D: col=l&15, row=(l>>4)*4+r.
// ---------------------------------------------------------------------------

template <int FM, int FN, int BK, typename T16>
struct BWaveTile {
  f32x4 acc[FM][FN];
  __device__ __forceinline__ void zero() {
#pragma unroll
    for (int i = 0; i < FM; ++i)
#pragma unroll
      for (int j = 0; j < FN; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};
  }
  __device__ __forceinline__ void mma(const T16* lds_a, const T16* lds_b,
                                      int wm, int wn, int lane) {
    constexpr int LDW = BK + BGEMM_PAD;
    const int fi = lane & 15;
    const int fk = (lane >> 4) * 8;
#pragma unroll
    for (int s = 0; s < BK / 32; ++s) {
      const int k = s * 32 + fk;
      using v8 = typename Vec16<T16>::v8;
      v8 a[FM], b[FN];
#pragma unroll
      for (int mi = 0; mi < FM; ++mi)
        a[mi] = *reinterpret_cast<const v8*>(
            &lds_a[(wm + mi * 16 + fi) * LDW + k]);
#pragma unroll
      for (int ni = 0; ni < FN; ++ni)
        b[ni] = *reinterpret_cast<const v8*>(
            &lds_b[(wn + ni * 16 + fi) * LDW + k]);
#pragma unroll
      for (int mi = 0; mi < FM; ++mi)
#pragma unroll
        for (int ni = 0; ni < FN; ++ni)
          acc[mi][ni] = Vec16<T16>::mfma(a[mi], b[ni], acc[mi][ni]);
    }
  }
  template <typename TO>
  __device__ __forceinline__ void store(TO* __restrict__ c, long ldc,
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
            c[(long)row * ldc + col] = (TO)v;
          }
        }
  }
};

// ---------------------------------------------------------------------------
// kernels — same three layouts as the f32 suite
// ---------------------------------------------------------------------------

template <int BM, int BN, int BK = 64, typename T16 = __bf16,
          typename TA = float, typename TO = float,
          int THREADS = BGEMM_THREADS>
__launch_bounds__(THREADS)
__global__ void gemm_bf16_nt_kernel(const TA* __restrict__ a,
                                    const float* __restrict__ b,
                                    const float* __restrict__ bias,
                                    TO* __restrict__ c, int m, int n, int k,
                                    int relu) {
  constexpr int LDW = BK + BGEMM_PAD;
  constexpr int WCOL = THREADS / PERTGNN_WAVE / 2;  // wave columns
  constexpr int FM = (BM / 2) / 16, FN = (BN / WCOL) / 16;
  __shared__ T16 lds_a[2][BM * LDW];
  __shared__ T16 lds_b[2][BN * LDW];
  const int bid = xcd_swizzle(blockIdx.x, gridDim.x);
  const int tiles_n = (n + BN - 1) / BN;
  const int m0 = (bid / tiles_n) * BM;
  const int n0 = (bid % tiles_n) * BN;
  const int wave = threadIdx.x / PERTGNN_WAVE;
  const int lane = threadIdx.x % PERTGNN_WAVE;
  const int wm = (wave / WCOL) * (BM / 2);
  const int wn = (wave % WCOL) * (BN / WCOL);

  BWaveTile<FM, FN, BK, T16> wt;
  wt.zero();
  int buf = 0;
  bstage_cmin<BM, BK, T16, THREADS>(a, k, m0, 0, m, k, lds_a[0]);
  bstage_cmin<BN, BK, T16, THREADS>(b, k, n0, 0, n, k, lds_b[0]);
  __syncthreads();
  StageCmin<BM, BK, T16, TA, THREADS> sa;
  StageCmin<BN, BK, T16, float, THREADS> sb;
  for (int k0 = BK; k0 < k; k0 += BK) {
    sa.load(a, k, m0, k0, m, k);        // issue loads…
    sb.load(b, k, n0, k0, n, k);
    wt.mma(lds_a[buf], lds_b[buf], wm, wn, lane);  // …MFMA while in flight
    sa.commit(lds_a[buf ^ 1]);
    sb.commit(lds_b[buf ^ 1]);
    __syncthreads();
    buf ^= 1;
  }
  wt.mma(lds_a[buf], lds_b[buf], wm, wn, lane);
  wt.store(c, n, m0 + wm, n0 + wn, m, n, bias, relu, lane);
}

template <int BM, int BN, int BK = 64, typename T16 = __bf16,
          typename TA = float, typename TO = float,
          int THREADS = BGEMM_THREADS>
__launch_bounds__(THREADS, 3)
__global__ void gemm_bf16_nn_kernel(const TA* __restrict__ a,
                                    const float* __restrict__ b,
                                    const float* __restrict__ bias,
                                    TO* __restrict__ c, int m, int n,
                                    int k2, int relu) {
  constexpr int LDW = BK + BGEMM_PAD;
  constexpr int WCOL = THREADS / PERTGNN_WAVE / 2;
  constexpr int FM = (BM / 2) / 16, FN = (BN / WCOL) / 16;
  __shared__ T16 lds_a[2][BM * LDW];
  __shared__ T16 lds_b[2][BN * LDW];
  const int bid = xcd_swizzle(blockIdx.x, gridDim.x);
  const int tiles_n = (k2 + BN - 1) / BN;
  const int m0 = (bid / tiles_n) * BM;
  const int n0 = (bid % tiles_n) * BN;
  const int wave = threadIdx.x / PERTGNN_WAVE;
  const int lane = threadIdx.x % PERTGNN_WAVE;
  const int wm = (wave / WCOL) * (BM / 2);
  const int wn = (wave % WCOL) * (BN / WCOL);

  BWaveTile<FM, FN, BK, T16> wt;
  wt.zero();
  int buf = 0;
  bstage_cmin<BM, BK, T16, THREADS>(a, n, m0, 0, m, n, lds_a[0]);
  bstage_cmaj<BN, BK, T16, THREADS>(b, k2, 0, n0, n, k2, lds_b[0]);
  __syncthreads();
  StageCmin<BM, BK, T16, TA, THREADS> sa;
  StageCmaj<BN, BK, T16, float, THREADS> sb;
  for (int c0 = BK; c0 < n; c0 += BK) {
    sa.load(a, n, m0, c0, m, n);
    sb.load(b, k2, c0, n0, n, k2);
    wt.mma(lds_a[buf], lds_b[buf], wm, wn, lane);
    sa.commit(lds_a[buf ^ 1]);
    sb.commit(lds_b[buf ^ 1]);
    __syncthreads();
    buf ^= 1;
  }
  wt.mma(lds_a[buf], lds_b[buf], wm, wn, lane);
  wt.store(c, k2, m0 + wm, n0 + wn, m, k2, bias, relu, lane);
}

template <int BM, int BN, int BK = 64, typename T16 = __bf16,
          typename TA = float, typename TB = float,
          int THREADS = BGEMM_THREADS>
__launch_bounds__(THREADS)
__global__ void gemm_bf16_tn_kernel(const TA* __restrict__ a,
                                    const TB* __restrict__ b,
                                    float* __restrict__ c,
                                    float* __restrict__ dbias, int m, int n,
                                    int k2, int slices) {
  constexpr int LDW = BK + BGEMM_PAD;
  constexpr int WCOL = THREADS / PERTGNN_WAVE / 2;
  constexpr int FM = (BM / 2) / 16, FN = (BN / WCOL) / 16;
  __shared__ T16 lds_a[2][BM * LDW];
  __shared__ T16 lds_b[2][BN * LDW];
  const int bid = xcd_swizzle(blockIdx.x, gridDim.x);
  const int tiles_k = (k2 + BN - 1) / BN;
  const int tile_id = bid / slices;
  const int slice = bid % slices;
  const int n0 = (tile_id / tiles_k) * BM;
  const int k0 = (tile_id % tiles_k) * BN;
  if (n0 >= n) return;
  const int wave = threadIdx.x / PERTGNN_WAVE;
  const int lane = threadIdx.x % PERTGNN_WAVE;
  const int wm = (wave / WCOL) * (BM / 2);
  const int wn = (wave % WCOL) * (BN / WCOL);

  const int per_slice =
      ((m + slices - 1) / slices + BK - 1) / BK * BK;
  const int c_beg = slice * per_slice;
  const int c_end = min(m, c_beg + per_slice);
  if (c_beg >= c_end) return;

  BWaveTile<FM, FN, BK, T16> wt;
  wt.zero();
  // fused bias grad: sum the STAGED bf16 A (=g) tile columns.  NOTE: this
  // sums bf16-rounded g — for exact-f32 db the caller uses the f32 path.
  const bool do_bias = (dbias != nullptr) && (k0 == 0);
  float dbsum = 0.f;
  const int bcol = threadIdx.x;
  int buf = 0;
  bstage_cmaj<BM, BK, T16, THREADS>(a, n, c_beg, n0, c_end, n, lds_a[0]);
  bstage_cmaj<BN, BK, T16, THREADS>(b, k2, c_beg, k0, c_end, k2, lds_b[0]);
  __syncthreads();
  StageCmaj<BM, BK, T16, TA, THREADS> sa;
  StageCmaj<BN, BK, T16, TB, THREADS> sb;
  for (int cc = c_beg + BK; cc < c_end; cc += BK) {
    sa.load(a, n, cc, n0, c_end, n);
    sb.load(b, k2, cc, k0, c_end, k2);
    wt.mma(lds_a[buf], lds_b[buf], wm, wn, lane);
    if (do_bias && bcol < BM)
#pragma unroll
      for (int r = 0; r < BK; ++r)
        dbsum += (float)lds_a[buf][bcol * LDW + r];
    sa.commit(lds_a[buf ^ 1]);
    sb.commit(lds_b[buf ^ 1]);
    __syncthreads();
    buf ^= 1;
  }
  wt.mma(lds_a[buf], lds_b[buf], wm, wn, lane);
  if (do_bias && bcol < BM) {
#pragma unroll
    for (int r = 0; r < BK; ++r)
      dbsum += (float)lds_a[buf][bcol * LDW + r];
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

// ---------------------------------------------------------------------------
// launchers
// ---------------------------------------------------------------------------

static int gemm_bk() {
  static int bk = [] {
    const char* e = getenv("PERTGNN_GEMM_BK");
    return (e && atoi(e) == 64) ? 64 : 32;  // default 32 (4 blocks/CU)
  }();
  return bk;
}

static bool gemm_t512() {  // 8-wave (512-thread) blocks for the a16 GEMMs
  static bool v = [] {
    const char* e = getenv("PERTGNN_GEMM_T512");
    return e && atoi(e) == 1;
  }();
  return v;
}

static bool gemm_n64() {  // narrow-BN experiment: 128x64 tiles, 4 waves/SIMD
  static bool v = [] {
    const char* e = getenv("PERTGNN_GEMM_N64");
    return e && atoi(e) == 1;
  }();
  return v;
}

void launch_gemm_bf16_nt(const float* a, const float* b, const float* bias,
                         float* c, int m, int n, int k, bool relu,
                         hipStream_t s) {
  if (m >= 512 && n >= 128) {
    if (gemm_n64()) {
      const int grid = ((m + 127) / 128) * ((n + 63) / 64);
      gemm_bf16_nt_kernel<128, 64, 32><<<dim3(grid), dim3(BGEMM_THREADS), 0, s>>>(
          a, b, bias, c, m, n, k, relu ? 1 : 0);
      return;
    }
    const int grid = ((m + 127) / 128) * ((n + 127) / 128);
    if (gemm_bk() == 32)
      gemm_bf16_nt_kernel<128, 128, 32><<<dim3(grid), dim3(BGEMM_THREADS), 0, s>>>(
          a, b, bias, c, m, n, k, relu ? 1 : 0);
    else
      gemm_bf16_nt_kernel<128, 128, 64><<<dim3(grid), dim3(BGEMM_THREADS), 0, s>>>(
          a, b, bias, c, m, n, k, relu ? 1 : 0);
  } else {
    const int grid = ((m + 63) / 64) * ((n + 63) / 64);
    gemm_bf16_nt_kernel<64, 64><<<dim3(grid), dim3(BGEMM_THREADS), 0, s>>>(
        a, b, bias, c, m, n, k, relu ? 1 : 0);
  }
}

void launch_gemm_bf16_nn(const float* a, const float* b, const float* bias,
                         float* c, int m, int n, int k2, bool relu,
                         hipStream_t s) {
  if (m >= 512 && k2 >= 128) {
    if (gemm_n64()) {
      const int grid = ((m + 127) / 128) * ((k2 + 63) / 64);
      gemm_bf16_nn_kernel<128, 64, 32><<<dim3(grid), dim3(BGEMM_THREADS), 0, s>>>(
          a, b, bias, c, m, n, k2, relu ? 1 : 0);
      return;
    }
    const int grid = ((m + 127) / 128) * ((k2 + 127) / 128);
    if (gemm_bk() == 32)
      gemm_bf16_nn_kernel<128, 128, 32><<<dim3(grid), dim3(BGEMM_THREADS), 0, s>>>(
          a, b, bias, c, m, n, k2, relu ? 1 : 0);
    else
      gemm_bf16_nn_kernel<128, 128, 64><<<dim3(grid), dim3(BGEMM_THREADS), 0, s>>>(
          a, b, bias, c, m, n, k2, relu ? 1 : 0);
  } else {
    const int grid = ((m + 63) / 64) * ((k2 + 63) / 64);
    gemm_bf16_nn_kernel<64, 64><<<dim3(grid), dim3(BGEMM_THREADS), 0, s>>>(
        a, b, bias, c, m, n, k2, relu ? 1 : 0);
  }
}

void launch_gemm_bf16_tn(const float* a, const float* b, float* c,
                         float* dbias, int m, int n, int k2, hipStream_t s) {
  const bool big = (n >= 128 && k2 >= 128);
  const int bm = big ? 128 : 64;
  const int bn = big ? 128 : 64;
  const int tiles = ((n + bm - 1) / bm) * ((k2 + bn - 1) / bn);
  int slices = 1;
  while (!pertgnn_deterministic() && tiles * slices < 512 && slices < 64 &&
         (long)slices * 64 * 4 < m)
    slices *= 2;
  if (slices > 1)
    HIP_CHECK(hipMemsetAsync(c, 0, (long)n * k2 * sizeof(float), s));
  if (dbias) HIP_CHECK(hipMemsetAsync(dbias, 0, n * sizeof(float), s));
  if (big) {
    if (gemm_bk() == 32)
      gemm_bf16_tn_kernel<128, 128, 32>
          <<<dim3(tiles * slices), dim3(BGEMM_THREADS), 0, s>>>(a, b, c, dbias,
                                                                m, n, k2,
                                                                slices);
    else
      gemm_bf16_tn_kernel<128, 128, 64>
          <<<dim3(tiles * slices), dim3(BGEMM_THREADS), 0, s>>>(a, b, c, dbias,
                                                                m, n, k2,
                                                                slices);
  } else
    gemm_bf16_tn_kernel<64, 64>
        <<<dim3(tiles * slices), dim3(BGEMM_THREADS), 0, s>>>(a, b, c, dbias,
                                                              m, n, k2,
                                                              slices);
}


// ---------------------------------------------------------------------------
// fp16 launchers (BASELINE config 5 — mfma_f32_16x16x32_f16, fp32 accumulate)
// ---------------------------------------------------------------------------

void launch_gemm_fp16_nt(const float* a, const float* b, const float* bias,
                         float* c, int m, int n, int k, bool relu,
                         hipStream_t s) {
  if (m >= 512 && n >= 128) {
    const int grid = ((m + 127) / 128) * ((n + 127) / 128);
    gemm_bf16_nt_kernel<128, 128, 32, _Float16>
        <<<dim3(grid), dim3(BGEMM_THREADS), 0, s>>>(a, b, bias, c, m, n, k,
                                                    relu ? 1 : 0);
  } else {
    const int grid = ((m + 63) / 64) * ((n + 63) / 64);
    gemm_bf16_nt_kernel<64, 64, 64, _Float16>
        <<<dim3(grid), dim3(BGEMM_THREADS), 0, s>>>(a, b, bias, c, m, n, k,
                                                    relu ? 1 : 0);
  }
}

void launch_gemm_fp16_nn(const float* a, const float* b, const float* bias,
                         float* c, int m, int n, int k2, bool relu,
                         hipStream_t s) {
  if (m >= 512 && k2 >= 128) {
    const int grid = ((m + 127) / 128) * ((k2 + 127) / 128);
    gemm_bf16_nn_kernel<128, 128, 32, _Float16>
        <<<dim3(grid), dim3(BGEMM_THREADS), 0, s>>>(a, b, bias, c, m, n, k2,
                                                    relu ? 1 : 0);
  } else {
    const int grid = ((m + 63) / 64) * ((k2 + 63) / 64);
    gemm_bf16_nn_kernel<64, 64, 64, _Float16>
        <<<dim3(grid), dim3(BGEMM_THREADS), 0, s>>>(a, b, bias, c, m, n, k2,
                                                    relu ? 1 : 0);
  }
}

void launch_gemm_fp16_tn(const float* a, const float* b, float* c,
                         float* dbias, int m, int n, int k2, hipStream_t s) {
  const bool big = (n >= 128 && k2 >= 128);
  const int bm = big ? 128 : 64;
  const int bn = big ? 128 : 64;
  const int tiles = ((n + bm - 1) / bm) * ((k2 + bn - 1) / bn);
  int slices = 1;
  while (!pertgnn_deterministic() && tiles * slices < 512 && slices < 64 &&
         (long)slices * 64 * 4 < m)
    slices *= 2;
  if (slices > 1)
    HIP_CHECK(hipMemsetAsync(c, 0, (long)n * k2 * sizeof(float), s));
  if (dbias) HIP_CHECK(hipMemsetAsync(dbias, 0, n * sizeof(float), s));
  if (big)
    gemm_bf16_tn_kernel<128, 128, 32, _Float16>
        <<<dim3(tiles * slices), dim3(BGEMM_THREADS), 0, s>>>(a, b, c, dbias,
                                                              m, n, k2,
                                                              slices);
  else
    gemm_bf16_tn_kernel<64, 64, 64, _Float16>
        <<<dim3(tiles * slices), dim3(BGEMM_THREADS), 0, s>>>(a, b, c, dbias,
                                                              m, n, k2,
                                                              slices);
}


// ---------------------------------------------------------------------------
// mixed-dtype launchers for the bf16-resident-activation mode: the QKVS
// forward writes its C as bf16; the backward GEMMs read the bf16 gradient
// directly (no conversion pass — bf16 LDS copy).
// ---------------------------------------------------------------------------

void launch_gemm_bf16_nt_o16(const float* a, const float* b, const float* bias,
                             void* c_v, int m, int n, int k, hipStream_t s) {
  __bf16* c = (__bf16*)c_v;
  if (m >= 512 && n >= 128) {
    const int grid = ((m + 127) / 128) * ((n + 127) / 128);
    gemm_bf16_nt_kernel<128, 128, 32, __bf16, float, __bf16>
        <<<dim3(grid), dim3(BGEMM_THREADS), 0, s>>>(a, b, bias, c, m, n, k, 0);
  } else {
    const int grid = ((m + 63) / 64) * ((n + 63) / 64);
    gemm_bf16_nt_kernel<64, 64, 64, __bf16, float, __bf16>
        <<<dim3(grid), dim3(BGEMM_THREADS), 0, s>>>(a, b, bias, c, m, n, k, 0);
  }
}

void launch_gemm_bf16_nn_a16(const void* a_v, const float* b, float* c, int m,
                             int n, int k2, hipStream_t s) {
  const __bf16* a = (const __bf16*)a_v;
  if (m >= 512 && k2 >= 128) {
    const int grid = ((m + 127) / 128) * ((k2 + 127) / 128);
    gemm_bf16_nn_kernel<128, 128, 32, __bf16, __bf16, float>
        <<<dim3(grid), dim3(BGEMM_THREADS), 0, s>>>(a, b, nullptr, c, m, n, k2,
                                                    0);
  } else {
    const int grid = ((m + 63) / 64) * ((k2 + 63) / 64);
    gemm_bf16_nn_kernel<64, 64, 64, __bf16, __bf16, float>
        <<<dim3(grid), dim3(BGEMM_THREADS), 0, s>>>(a, b, nullptr, c, m, n, k2,
                                                    0);
  }
}

void launch_gemm_bf16_tn_a16(const void* a_v, const float* b, float* c,
                             float* dbias, int m, int n, int k2,
                             hipStream_t s) {
  const __bf16* a = (const __bf16*)a_v;
  const bool big = (n >= 128 && k2 >= 128);
  const int bm = big ? 128 : 64;
  const int bn = big ? 128 : 64;
  const int tiles = ((n + bm - 1) / bm) * ((k2 + bn - 1) / bn);
  int slices = 1;
  while (!pertgnn_deterministic() && tiles * slices < 512 && slices < 64 &&
         (long)slices * 64 * 4 < m)
    slices *= 2;
  if (slices > 1)
    HIP_CHECK(hipMemsetAsync(c, 0, (long)n * k2 * sizeof(float), s));
  if (dbias) HIP_CHECK(hipMemsetAsync(dbias, 0, n * sizeof(float), s));
  if (big)
    gemm_bf16_tn_kernel<128, 128, 32, __bf16, __bf16>
        <<<dim3(tiles * slices), dim3(BGEMM_THREADS), 0, s>>>(a, b, c, dbias,
                                                              m, n, k2,
                                                              slices);
  else
    gemm_bf16_tn_kernel<64, 64, 64, __bf16, __bf16>
        <<<dim3(tiles * slices), dim3(BGEMM_THREADS), 0, s>>>(a, b, c, dbias,
                                                              m, n, k2,
                                                              slices);
}

// ---------------------------------------------------------------------------
// act16-v2 launchers: fully bf16 activation IO (x16 in / qkvs16 out on the
// forward, bf16 gradients through dgrad, bf16 x through wgrad).  Weights and
// accumulation stay fp32; only stream dtypes change — at the flagship shapes
// the three model GEMMs are stream-bound, so halving A/C bytes is the lever
// (isolated: hipBLASLt bf16 reaches 1.9-3.5 TB/s effective on these shapes).
// ---------------------------------------------------------------------------

void launch_gemm_bf16_nt_a16o16(const void* a_v, const float* b,
                                const float* bias, void* c_v, int m, int n,
                                int k, hipStream_t s) {
  const __bf16* a = (const __bf16*)a_v;
  __bf16* c = (__bf16*)c_v;
  if (m >= 512 && n >= 128) {
    const int grid = ((m + 127) / 128) * ((n + 127) / 128);
    if (gemm_t512()) {
      gemm_bf16_nt_kernel<128, 128, 32, __bf16, __bf16, __bf16, 512>
          <<<dim3(grid), dim3(512), 0, s>>>(a, b, bias, c, m, n, k, 0);
      return;
    }
    gemm_bf16_nt_kernel<128, 128, 32, __bf16, __bf16, __bf16>
        <<<dim3(grid), dim3(BGEMM_THREADS), 0, s>>>(a, b, bias, c, m, n, k, 0);
  } else {
    const int grid = ((m + 63) / 64) * ((n + 63) / 64);
    gemm_bf16_nt_kernel<64, 64, 64, __bf16, __bf16, __bf16>
        <<<dim3(grid), dim3(BGEMM_THREADS), 0, s>>>(a, b, bias, c, m, n, k, 0);
  }
}

void launch_gemm_bf16_nn_a16o16(const void* a_v, const float* b, void* c_v,
                                int m, int n, int k2, hipStream_t s) {
  const __bf16* a = (const __bf16*)a_v;
  __bf16* c = (__bf16*)c_v;
  if (m >= 512 && k2 >= 128) {
    const int grid = ((m + 127) / 128) * ((k2 + 127) / 128);
    if (gemm_t512()) {
      gemm_bf16_nn_kernel<128, 128, 32, __bf16, __bf16, __bf16, 512>
          <<<dim3(grid), dim3(512), 0, s>>>(a, b, nullptr, c, m, n, k2, 0);
      return;
    }
    gemm_bf16_nn_kernel<128, 128, 32, __bf16, __bf16, __bf16>
        <<<dim3(grid), dim3(BGEMM_THREADS), 0, s>>>(a, b, nullptr, c, m, n, k2,
                                                    0);
  } else {
    const int grid = ((m + 63) / 64) * ((k2 + 63) / 64);
    gemm_bf16_nn_kernel<64, 64, 64, __bf16, __bf16, __bf16>
        <<<dim3(grid), dim3(BGEMM_THREADS), 0, s>>>(a, b, nullptr, c, m, n, k2,
                                                    0);
  }
}

void launch_gemm_bf16_tn_a16b16(const void* a_v, const void* b_v, float* c,
                                float* dbias, int m, int n, int k2,
                                hipStream_t s) {
  const __bf16* a = (const __bf16*)a_v;
  const __bf16* b = (const __bf16*)b_v;
  const bool big = (n >= 128 && k2 >= 128);
  const int bm = big ? 128 : 64;
  const int bn = big ? 128 : 64;
  const int tiles = ((n + bm - 1) / bm) * ((k2 + bn - 1) / bn);
  int slices = 1;
  while (!pertgnn_deterministic() && tiles * slices < 512 && slices < 64 &&
         (long)slices * 64 * 4 < m)
    slices *= 2;
  if (slices > 1)
    HIP_CHECK(hipMemsetAsync(c, 0, (long)n * k2 * sizeof(float), s));
  if (dbias) HIP_CHECK(hipMemsetAsync(dbias, 0, n * sizeof(float), s));
  if (big) {
    if (gemm_t512()) {
      gemm_bf16_tn_kernel<128, 128, 32, __bf16, __bf16, __bf16, 512>
          <<<dim3(tiles * slices), dim3(512), 0, s>>>(a, b, c, dbias, m, n,
                                                      k2, slices);
      return;
    }
    gemm_bf16_tn_kernel<128, 128, 32, __bf16, __bf16, __bf16>
        <<<dim3(tiles * slices), dim3(BGEMM_THREADS), 0, s>>>(a, b, c, dbias,
                                                              m, n, k2,
                                                              slices);
  } else
    gemm_bf16_tn_kernel<64, 64, 64, __bf16, __bf16, __bf16>
        <<<dim3(tiles * slices), dim3(BGEMM_THREADS), 0, s>>>(a, b, c, dbias,
                                                              m, n, k2,
                                                              slices);
}

// fp16-COMPUTE variants of the act16 launchers: activation streams stay bf16
// (storage dtype is independent of the MFMA operand dtype — staging converts
// while packing), matrix cores run v_mfma_f32_16x16x32_f16.
void launch_gemm_fp16_nt_a16o16(const void* a_v, const float* b,
                                const float* bias, void* c_v, int m, int n,
                                int k, hipStream_t s) {
  const __bf16* a = (const __bf16*)a_v;
  __bf16* c = (__bf16*)c_v;
  if (m >= 512 && n >= 128) {
    const int grid = ((m + 127) / 128) * ((n + 127) / 128);
    gemm_bf16_nt_kernel<128, 128, 32, _Float16, __bf16, __bf16>
        <<<dim3(grid), dim3(BGEMM_THREADS), 0, s>>>(a, b, bias, c, m, n, k, 0);
  } else {
    const int grid = ((m + 63) / 64) * ((n + 63) / 64);
    gemm_bf16_nt_kernel<64, 64, 64, _Float16, __bf16, __bf16>
        <<<dim3(grid), dim3(BGEMM_THREADS), 0, s>>>(a, b, bias, c, m, n, k, 0);
  }
}

void launch_gemm_fp16_nn_a16o16(const void* a_v, const float* b, void* c_v,
                                int m, int n, int k2, hipStream_t s) {
  const __bf16* a = (const __bf16*)a_v;
  __bf16* c = (__bf16*)c_v;
  if (m >= 512 && k2 >= 128) {
    const int grid = ((m + 127) / 128) * ((k2 + 127) / 128);
    gemm_bf16_nn_kernel<128, 128, 32, _Float16, __bf16, __bf16>
        <<<dim3(grid), dim3(BGEMM_THREADS), 0, s>>>(a, b, nullptr, c, m, n, k2,
                                                    0);
  } else {
    const int grid = ((m + 63) / 64) * ((k2 + 63) / 64);
    gemm_bf16_nn_kernel<64, 64, 64, _Float16, __bf16, __bf16>
        <<<dim3(grid), dim3(BGEMM_THREADS), 0, s>>>(a, b, nullptr, c, m, n, k2,
                                                    0);
  }
}

void launch_gemm_fp16_tn_a16b16(const void* a_v, const void* b_v, float* c,
                                float* dbias, int m, int n, int k2,
                                hipStream_t s) {
  const __bf16* a = (const __bf16*)a_v;
  const __bf16* b = (const __bf16*)b_v;
  const bool big = (n >= 128 && k2 >= 128);
  const int bm = big ? 128 : 64;
  const int bn = big ? 128 : 64;
  const int tiles = ((n + bm - 1) / bm) * ((k2 + bn - 1) / bn);
  int slices = 1;
  while (!pertgnn_deterministic() && tiles * slices < 512 && slices < 64 &&
         (long)slices * 64 * 4 < m)
    slices *= 2;
  if (slices > 1)
    HIP_CHECK(hipMemsetAsync(c, 0, (long)n * k2 * sizeof(float), s));
  if (dbias) HIP_CHECK(hipMemsetAsync(dbias, 0, n * sizeof(float), s));
  if (big)
    gemm_bf16_tn_kernel<128, 128, 32, _Float16, __bf16, __bf16>
        <<<dim3(tiles * slices), dim3(BGEMM_THREADS), 0, s>>>(a, b, c, dbias,
                                                              m, n, k2,
                                                              slices);
  else
    gemm_bf16_tn_kernel<64, 64, 64, _Float16, __bf16, __bf16>
        <<<dim3(tiles * slices), dim3(BGEMM_THREADS), 0, s>>>(a, b, c, dbias,
                                                              m, n, k2,
                                                              slices);
}

// ---------------------------------------------------------------------------
// glds NT kernel: both operands bf16 in HBM, staged by global_load_lds
// width-16 (direct-to-LDS DMA — no staging VGPRs, no ds_write pass; guide
// "optimization ladder" step 3: +67% over register staging on its own).
// LDS image is lane-linear row-major [128 rows][64 k] bf16 with the
// st_16x32 XOR swizzle (byte ^= ((byte>>9)&1)<<5 within each 1024-B
// subtile) applied on BOTH the per-lane glds SOURCE address and the
// ds_read address: linear 128-B rows put every fragment lane-group on the
// same bank row (8-way conflict); the swizzle spreads rows 4..7 of each
// 8-row group one 32-B slot over (guide: bank-conflict 141x down).
// One barrier per K-step: issue next tile's glds -> MFMA current ->
// s_waitcnt vmcnt(0) -> barrier (the accepted step-3 stall).
// Interior tiles only — the launcher routes edge tiles (m tail) to the
// register-staging kernel via pointer offset; requires k % BK == 0 and
// 16-B-aligned rows (lda % 8 == 0).
// ---------------------------------------------------------------------------

__device__ __forceinline__ unsigned glds_swz(unsigned byte_off) {
  return byte_off ^ (((byte_off >> 9) & 1u) << 5);
}

typedef __attribute__((ext_vector_type(4))) unsigned int u32x4_t;

template <int BM, int BN, int BK, typename TO, int THREADS, bool NTC>
__launch_bounds__(THREADS)
__global__ void gemm_a16_glds_nt_kernel(const __bf16* __restrict__ a,
                                        const __bf16* __restrict__ b,
                                        const float* __restrict__ bias,
                                        TO* __restrict__ c, int m, int n,
                                        int k, int relu) {
  constexpr int NWAVE = THREADS / PERTGNN_WAVE;
  constexpr int WCOL = NWAVE / 2;
  constexpr int FM = (BM / 2) / 16, FN = (BN / WCOL) / 16;
  constexpr int TILE_BYTES = BM * BK * 2;          // A-operand K-tile
  constexpr int GRPS_A = TILE_BYTES / 1024;        // 1 KiB per wave-glds
  constexpr int GRPS_B = (BN * BK * 2) / 1024;
  // one arena: [A0 A1 B0 B1] operand buffers; the C-store scratch reuses it
  __shared__ char smem[2 * (BM + BN) * BK * 2];
  const auto lds_ab = [&](int i) { return (__bf16*)(smem + i * TILE_BYTES); };
  const auto lds_bb = [&](int i) {
    return (__bf16*)(smem + 2 * TILE_BYTES + i * (BN * BK * 2));
  };
  const int bid = xcd_swizzle(blockIdx.x, gridDim.x);
  const int tiles_n = n / BN;
  const int m0 = (bid / tiles_n) * BM;
  const int n0 = (bid % tiles_n) * BN;
  const int wave = threadIdx.x / PERTGNN_WAVE;
  const int lane = threadIdx.x % PERTGNN_WAVE;
  const int wm = (wave / WCOL) * (BM / 2);
  const int wn = (wave % WCOL) * (BN / WCOL);

  // per-lane glds source mapping: lds slot (grp, lane*16) holds the element
  // at LOGICAL tile offset swz(grp*1024 + lane*16) = (row, kbyte); row
  // width is BK*2 bytes, so one 1-KiB group covers 1024/(BK*2) rows
  constexpr int ROW_BYTES = BK * 2;
  constexpr int ROWS_PER_GRP = 1024 / ROW_BYTES;
  const unsigned l_off = glds_swz((unsigned)lane * 16);
  const int src_row = (int)(l_off / ROW_BYTES);    // within the group
  const int src_kb = (int)(l_off % ROW_BYTES);     // byte within row

  const long lda = k;  // elements
  // per-operand group counts: A covers BM rows, B covers BN (they differ
  // for the single-column-tile dgrad variant)
  auto stage = [&]<int GRPS>(const __bf16* g, int g0, int k0, __bf16* lds) {
#pragma unroll
    for (int i = 0; i < GRPS / NWAVE; ++i) {
      const int grp = wave + i * NWAVE;
      const __bf16* src = (const __bf16*)((const char*)g +
          (long)(g0 + grp * ROWS_PER_GRP + src_row) * lda * 2 +
          (long)k0 * 2 + src_kb);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)src,
          (__attribute__((address_space(3))) void*)(lds + grp * 512),
          16, 0, 0);
    }
  };

  f32x4 acc[FM][FN];
#pragma unroll
  for (int i = 0; i < FM; ++i)
#pragma unroll
    for (int j = 0; j < FN; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  stage.template operator()<GRPS_A>(a, m0, 0, lds_ab(0));
  stage.template operator()<GRPS_B>(b, n0, 0, lds_bb(0));
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  const int fi = lane & 15;
  const int fk = (lane >> 4) * 8;
  auto mma = [&](const __bf16* la, const __bf16* lb) {
#pragma unroll
    for (int s = 0; s < BK / 32; ++s) {
      const int kb = (s * 32 + fk) * 2;
      bf16x8 av[FM], bv[FN];
#pragma unroll
      for (int mi = 0; mi < FM; ++mi) {
        const int row = wm + mi * 16 + fi;
        av[mi] = *reinterpret_cast<const bf16x8*>(
            (const char*)la + glds_swz((unsigned)(row * (BK * 2) + kb)));
      }
#pragma unroll
      for (int ni = 0; ni < FN; ++ni) {
        const int row = wn + ni * 16 + fi;
        bv[ni] = *reinterpret_cast<const bf16x8*>(
            (const char*)lb + glds_swz((unsigned)(row * (BK * 2) + kb)));
      }
#pragma unroll
      for (int mi = 0; mi < FM; ++mi)
#pragma unroll
        for (int ni = 0; ni < FN; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              av[mi], bv[ni], acc[mi][ni], 0, 0, 0);
    }
  };

  int buf = 0;
  for (int k0 = BK; k0 < k; k0 += BK) {
    stage.template operator()<GRPS_A>(a, m0, k0, lds_ab(buf ^ 1));
    stage.template operator()<GRPS_B>(b, n0, k0, lds_bb(buf ^ 1));
    mma(lds_ab(buf), lds_bb(buf));
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    buf ^= 1;
  }
  mma(lds_ab(buf), lds_bb(buf));

  // LDS-staged coalesced epilogue: the MFMA C fragment layout (lane owns
  // col=l&15, rows (l>>4)*4+r) would store 2-B/4-B scalars at row stride —
  // the measured bottleneck at [M,1024] outputs.  Each wave transposes one
  // 16-row band at a time through its private LDS scratch, then writes
  // 16-B lane chunks (8 lanes cover a full 128-B row of bf16).  NTC stores
  // the C stream nontemporally — 370 MB of writes with zero reuse would
  // otherwise wash the operand tiles out of L2.
  __syncthreads();  // operand LDS is being re-purposed as scratch
  constexpr int BNW = BN / WCOL;              // wave's output columns
  constexpr int SW = BNW + 8;                 // padded scratch stride
  TO* scratch = reinterpret_cast<TO*>(smem + wave * 16 * SW * sizeof(TO));
  const int fcol = lane & 15;
  const int frow = (lane >> 4) * 4;
  constexpr int EPL = 16 / sizeof(TO);        // elements per 16-B chunk
  constexpr int CPR = BNW / EPL;              // chunks per scratch row
#pragma unroll
  for (int mi = 0; mi < FM; ++mi) {
#pragma unroll
    for (int ni = 0; ni < FN; ++ni)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int col = ni * 16 + fcol;
        float v = acc[mi][ni][r];
        if (bias) v += bias[n0 + wn + col];
        if (relu) v = fmaxf(v, 0.f);
        scratch[(frow + r) * SW + col] = (TO)v;
      }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
#pragma unroll
    for (int it = 0; it < 16 * CPR / PERTGNN_WAVE; ++it) {
      const int chunk = it * PERTGNN_WAVE + lane;
      const int r = chunk / CPR;
      const int off = (chunk % CPR) * EPL;
      const u32x4_t val = *reinterpret_cast<const u32x4_t*>(&scratch[r * SW + off]);
      u32x4_t* dst = reinterpret_cast<u32x4_t*>(
          &c[(long)(m0 + wm + mi * 16 + r) * n + n0 + wn + off]);
      if constexpr (NTC)
        __builtin_nontemporal_store(val, dst);
      else
        *dst = val;
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  }
}

// fp32 [n][k] row-major -> bf16 [n][k] (weight operand for the glds path)
__global__ void convert_w16_kernel(const float* __restrict__ w,
                                   __bf16* __restrict__ o, long numel) {
  const long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long t = i0; t < numel; t += stride) o[t] = (__bf16)w[t];
}

// fp32 [rows][cols] -> bf16 [cols][rows] (transposed weight for dgrad-as-NT)
__global__ void transpose_convert_w16_kernel(const float* __restrict__ w,
                                             __bf16* __restrict__ o, int rows,
                                             int cols) {
  __shared__ float tile[32][33];
  const int c0 = blockIdx.x * 32;
  const int r0 = blockIdx.y * 32;
  const int tx = threadIdx.x % 32;
  const int ty = threadIdx.x / 32;
  for (int dy = ty; dy < 32; dy += blockDim.x / 32) {
    const int r = r0 + dy, ccol = c0 + tx;
    tile[dy][tx] = (r < rows && ccol < cols) ? w[(long)r * cols + ccol] : 0.f;
  }
  __syncthreads();
  for (int dy = ty; dy < 32; dy += blockDim.x / 32) {
    const int ccol = c0 + dy, r = r0 + tx;
    if (ccol < cols && r < rows) o[(long)ccol * rows + r] = (__bf16)tile[tx][dy];
  }
}

void launch_convert_w16(const float* w, void* o, long numel, hipStream_t s) {
  const int blocks = (int)min((numel + 255) / 256, (long)4096);
  convert_w16_kernel<<<blocks, 256, 0, s>>>(w, (__bf16*)o, numel);
}

void launch_transpose_convert_w16(const float* w, void* o, int rows, int cols,
                                  hipStream_t s) {
  transpose_convert_w16_kernel<<<dim3((cols + 31) / 32, (rows + 31) / 32),
                                 dim3(256), 0, s>>>(w, (__bf16*)o, rows, cols);
}

// A + B16 bf16, NT: full interior tiles through the glds kernel, the
// m-tail strip through the register-staging kernel (pointer offset; it
// stages the ORIGINAL fp32 weights, rounding to the same bf16 values the
// convert kernel produced).
static int glds_bk() {  // default 32 (measured: +13-20% over 64 at model shapes)
  static int v = [] {
    const char* e = getenv("PERTGNN_GLDS_BK");
    return (e && atoi(e) == 64) ? 64 : 32;
  }();
  return v;
}
static bool glds_t512() {  // default ON (8 waves; measured faster everywhere)
  static bool v = [] {
    const char* e = getenv("PERTGNN_GLDS_T512");
    return !(e && atoi(e) == 0);
  }();
  return v;
}
static bool glds_ntc() {  // nontemporal C stores (default ON)
  static bool v = [] {
    const char* e = getenv("PERTGNN_GLDS_NTC");
    return !(e && atoi(e) == 0);
  }();
  return v;
}


static bool glds_n256() {  // 256-wide output tile for big dgrads (default ON)
  static bool v = [] {
    const char* e = getenv("PERTGNN_GLDS_N256");
    return !(e && atoi(e) == 0);
  }();
  return v;
}

void launch_gemm_a16_glds_nt(const void* a_v, const void* b16_v,
                             const float* b32, const float* bias, void* c_v,
                             int c16, int m, int n, int k, bool relu,
                             hipStream_t s) {
  const __bf16* a = (const __bf16*)a_v;
  const __bf16* b = (const __bf16*)b16_v;
  constexpr int BM = 128, BN = 128;
  const int mt = m / BM;  // full row tiles
  if (mt >= 1024 && n == 256 && k % 32 == 0 && glds_n256()) {
    // single-column-tile variant for the dgrad shape: A is streamed exactly
    // once and the per-block K-loop halves its barrier count (measured
    // +12% at [180k,1024]x[1024,256]); needs mt >= ~1024 blocks to fill
    // the chip (it LOSES at 45k rows where the 128-wide grid is 2x larger)
    if (c16)
      gemm_a16_glds_nt_kernel<128, 256, 32, __bf16, 512, true>
          <<<dim3(mt), dim3(512), 0, s>>>(a, b, bias, (__bf16*)c_v, m, n, k,
                                          relu ? 1 : 0);
    else
      gemm_a16_glds_nt_kernel<128, 256, 32, float, 512, true>
          <<<dim3(mt), dim3(512), 0, s>>>(a, b, bias, (float*)c_v, m, n, k,
                                          relu ? 1 : 0);
  } else if (mt > 0) {
    const int grid = mt * (n / BN);
    const int bk = glds_bk();
    const bool t512 = glds_t512();
    const bool ntc = glds_ntc();
#define GLDS_DISPATCH(BKV, TH, NTCV)                                           \
  do {                                                                         \
    if (c16)                                                                   \
      gemm_a16_glds_nt_kernel<BM, BN, BKV, __bf16, TH, NTCV>                   \
          <<<dim3(grid), dim3(TH), 0, s>>>(a, b, bias, (__bf16*)c_v, m, n, k,  \
                                           relu ? 1 : 0);                      \
    else                                                                       \
      gemm_a16_glds_nt_kernel<BM, BN, BKV, float, TH, NTCV>                    \
          <<<dim3(grid), dim3(TH), 0, s>>>(a, b, bias, (float*)c_v, m, n, k,   \
                                           relu ? 1 : 0);                      \
  } while (0)
    if ((bk == 32 || k % 64 != 0) && k % 32 == 0) {
      if (t512) { if (ntc) GLDS_DISPATCH(32, 512, true); else GLDS_DISPATCH(32, 512, false); }
      else      { if (ntc) GLDS_DISPATCH(32, 256, true); else GLDS_DISPATCH(32, 256, false); }
    } else {
      if (t512) { if (ntc) GLDS_DISPATCH(64, 512, true); else GLDS_DISPATCH(64, 512, false); }
      else      { if (ntc) GLDS_DISPATCH(64, 256, true); else GLDS_DISPATCH(64, 256, false); }
    }
#undef GLDS_DISPATCH
  }
  const int m_done = mt * BM;
  if (m_done < m && b32) {  // tail strip via the register-staging kernel
    const int ms = m - m_done;
    const int grid = ((ms + 63) / 64) * ((n + 63) / 64);
    if (c16)
      gemm_bf16_nt_kernel<64, 64, 64, __bf16, __bf16, __bf16>
          <<<dim3(grid), dim3(BGEMM_THREADS), 0, s>>>(
              a + (long)m_done * k, b32, bias,
              (__bf16*)c_v + (long)m_done * n, ms, n, k, relu ? 1 : 0);
    else
      gemm_bf16_nt_kernel<64, 64, 64, __bf16, __bf16, float>
          <<<dim3(grid), dim3(BGEMM_THREADS), 0, s>>>(
              a + (long)m_done * k, b32, bias,
              (float*)c_v + (long)m_done * n, ms, n, k, relu ? 1 : 0);
  }
}

// ---------------------------------------------------------------------------
// glds TN kernel (wgrad): dw[n,k2] = sum_m g[m,n]^T x[m,k2].  Both operands
// are contract(m)-major in HBM, so the NT kernel's row-linear glds image
// cannot feed k-minor MFMA fragments.  Instead each operand tile is staged
// as PANEL-MAJOR images — 16-column panels of [BKM m][16 cols] row-major
// (the per-lane glds source addresses assemble them directly) — and the
// MFMA fragments are read with gfx950's ds_read_b64_tr_b16 hardware
// transpose-read (lane gets 4 contract-consecutive elements at 32-B
// stride; two reads make the 8-element k fragment).  Split-K over m with
// atomicAdd combine, same contract as the register-staging TN kernel.
// ---------------------------------------------------------------------------

typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4_g;

template <int BM /*n cols*/, int BN /*k2 cols*/, int BKM /*m per step*/,
          int THREADS = 512>
__launch_bounds__(THREADS)
__global__ void gemm_a16_glds_tn_kernel(const __bf16* __restrict__ a,
                                        const __bf16* __restrict__ b,
                                        float* __restrict__ c,
                                        float* __restrict__ dbias, int m,
                                        int n, int k2, int slices) {
  constexpr int NWAVE = THREADS / PERTGNN_WAVE;
  constexpr int WCOL = NWAVE / 2;
  constexpr int FM = (BM / 2) / 16, FN = (BN / WCOL) / 16;
  constexpr int PANEL_BYTES = BKM * 32;            // [BKM][16] bf16
  constexpr int TILE_BYTES = BM * BKM * 2;
  constexpr int GRPS = TILE_BYTES / 1024;          // 1 KiB per wave-glds
  constexpr int GRPS_PER_WAVE = GRPS / NWAVE;
  constexpr int ROWS_PER_GRP = 1024 / 32;          // 32 m-rows per group
  __shared__ char smem[2 * (BM + BN) * BKM * 2];
  const auto lds_ab = [&](int i) { return smem + i * TILE_BYTES; };
  const auto lds_bb = [&](int i) {
    return smem + 2 * TILE_BYTES + i * (BN * BKM * 2);
  };
  const int bid = xcd_swizzle(blockIdx.x, gridDim.x);
  const int tiles_k = k2 / BN;
  const int tile_id = bid / slices;
  const int slice = bid % slices;
  const int n0 = (tile_id / tiles_k) * BM;
  const int k0 = (tile_id % tiles_k) * BN;
  const int wave = threadIdx.x / PERTGNN_WAVE;
  const int lane = threadIdx.x % PERTGNN_WAVE;
  const int wm = (wave / WCOL) * (BM / 2);
  const int wn = (wave % WCOL) * (BN / WCOL);

  // m range of this slice (multiples of BKM; the launcher routes the
  // global m tail to the register-staging kernel)
  const int m_full = m - m % BKM;
  const int per_slice = ((m_full / BKM + slices - 1) / slices) * BKM;
  const int c_beg = slice * per_slice;
  const int c_end = min(m_full, c_beg + per_slice);
  if (c_beg >= c_end) return;

  // panel-major glds staging: group g covers half a panel (32 m x 16 cols);
  // lane's 16-B chunk = 8 columns of one m-row of the panel
  auto stage = [&](const __bf16* src_base, long ld, int col0, int mrow0,
                   char* lds) {
#pragma unroll
    for (int i = 0; i < GRPS_PER_WAVE; ++i) {
      const int grp = wave + i * NWAVE;
      const int panel = grp / (PANEL_BYTES / 1024);
      const int half = grp % (PANEL_BYTES / 1024);
      const int mrow = half * ROWS_PER_GRP + (lane >> 1);
      const int col = panel * 16 + (lane & 1) * 8;
      const __bf16* src =
          src_base + (long)(mrow0 + mrow) * ld + col0 + col;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)src,
          (__attribute__((address_space(3))) void*)(lds + grp * 1024),
          16, 0, 0);
    }
  };

  f32x4 acc[FM][FN];
  float dbacc[FM];  // fused bias grad rides the A fragments (see below)
#pragma unroll
  for (int i = 0; i < FM; ++i) {
    dbacc[i] = 0.f;
#pragma unroll
    for (int j = 0; j < FN; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};
  }

  stage(a, n, n0, c_beg, lds_ab(0));
  stage(b, k2, k0, c_beg, lds_bb(0));
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  const bool do_bias = (dbias != nullptr) && (k0 == 0);

  // fragment read via ds_read_b64_tr_b16: each 16-lane group passes
  // CONSECUTIVE 8-B addresses covering one 128-B [4 m][16 col] sub-tile;
  // the hardware redistributes so lane receives COLUMN (l&15) — i.e. 4
  // contract-consecutive elements.  Two reads (sub-tiles m+0..3, m+4..7)
  // assemble the 8-element MFMA k fragment.
  auto frag = [&](const char* img, int block16, int s) {
    const char* p = img + (long)block16 * PANEL_BYTES +
                    (s * 32 + ((lane >> 4) * 8)) * 32 + (lane & 15) * 8;
    bf16x4_g lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
        (__attribute__((address_space(3))) bf16x4_g*)p);
    bf16x4_g hi = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
        (__attribute__((address_space(3))) bf16x4_g*)(p + 128));
    union { bf16x8 v8; bf16x4_g v4[2]; } u;
    u.v4[0] = lo;
    u.v4[1] = hi;
    return u.v8;
  };

  // bias grad (db = colsum of g): the A fragments already pass EVERY g
  // element through registers, so accumulate there instead of a separate
  // 32-B-strided LDS sweep (PMC: that sweep was 8.5x the NT kernel's LDS
  // bank conflicts).  Each column's sum lands spread over the wave's four
  // 16-lane groups (different k slices) — folded by shfl at the end; the
  // WCOL waves sharing a wm row-block read identical fragments, so only
  // wave % WCOL == 0 contributes.
  auto mma = [&](const char* la, const char* lb) {
#pragma unroll
    for (int s = 0; s < BKM / 32; ++s) {
      bf16x8 av[FM], bv[FN];
#pragma unroll
      for (int mi = 0; mi < FM; ++mi) av[mi] = frag(la, (wm >> 4) + mi, s);
#pragma unroll
      for (int ni = 0; ni < FN; ++ni) bv[ni] = frag(lb, (wn >> 4) + ni, s);
      if (do_bias)
#pragma unroll
        for (int mi = 0; mi < FM; ++mi)
#pragma unroll
          for (int j = 0; j < 8; ++j) dbacc[mi] += (float)av[mi][j];
#pragma unroll
      for (int mi = 0; mi < FM; ++mi)
#pragma unroll
        for (int ni = 0; ni < FN; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              av[mi], bv[ni], acc[mi][ni], 0, 0, 0);
    }
  };

  int buf = 0;
  for (int cc = c_beg + BKM; cc < c_end; cc += BKM) {
    stage(a, n, n0, cc, lds_ab(buf ^ 1));
    stage(b, k2, k0, cc, lds_bb(buf ^ 1));
    mma(lds_ab(buf), lds_bb(buf));
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    buf ^= 1;
  }
  mma(lds_ab(buf), lds_bb(buf));
  if (do_bias && wave % WCOL == 0) {
#pragma unroll
    for (int mi = 0; mi < FM; ++mi) {
      float v = dbacc[mi];
      v += __shfl_xor(v, 16, 64);
      v += __shfl_xor(v, 32, 64);
      if ((lane >> 4) == 0)
        atomicAdd(&dbias[n0 + wm + mi * 16 + (lane & 15)], v);
    }
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
        if (slices == 1)
          c[(long)row * k2 + col] = acc[mi][ni][r];
        else
          atomicAdd(&c[(long)row * k2 + col], acc[mi][ni][r]);
      }
}

// m-tail of the TN wgrad: C[n,k2] += sum_{i<mt} g[i,n_idx]*x[i,k_idx] for the
// mt (< BKM) rows past the last full m tile.  The work is tiny (mt*n*k2
// MACs) but the register-staging kernel used to run it on a
// (n/128)*(k2/128)-block grid — 16-32 workgroups of guarded scalar staging,
// ~107 us for 0.01 GFLOP at [26,1024,256].  Here one block owns one output
// ROW of C (grid = n, fills the chip), threads stride k2: the x reads are
// lane-coalesced, g[i,jn] is a wave-broadcast scalar, and the contribution
// is atomicAdd-combined with the main kernel's slices exactly like an extra
// split-K slice.
__global__ void gemm_tail_tn_kernel(const __bf16* __restrict__ g,
                                    const __bf16* __restrict__ b,
                                    float* __restrict__ c,
                                    float* __restrict__ dbias, int mt, int n,
                                    int k2) {
  const int jn = blockIdx.x;
  if (dbias && threadIdx.x == 0) {
    float s = 0.f;
    for (int i = 0; i < mt; ++i) s += (float)g[(long)i * n + jn];
    atomicAdd(&dbias[jn], s);
  }
  for (int jk = threadIdx.x; jk < k2; jk += blockDim.x) {
    float a0 = 0.f, a1 = 0.f;
    int i = 0;
    for (; i + 2 <= mt; i += 2) {
      a0 += (float)g[(long)i * n + jn] * (float)b[(long)i * k2 + jk];
      a1 += (float)g[(long)(i + 1) * n + jn] * (float)b[(long)(i + 1) * k2 + jk];
    }
    if (i < mt)
      a0 += (float)g[(long)i * n + jn] * (float)b[(long)i * k2 + jk];
    atomicAdd(&c[(long)jn * k2 + jk], a0 + a1);
  }
}

// Both operands bf16 contract-major (g [m][n], x [m][k2]); m tail rows
// beyond the last BKM multiple run gemm_tail_tn_kernel as an extra atomic
// slice.
void launch_gemm_a16_glds_tn(const void* a_v, const void* b_v, float* c,
                             float* dbias, int m, int n, int k2,
                             hipStream_t s) {
  const __bf16* a = (const __bf16*)a_v;
  const __bf16* b = (const __bf16*)b_v;
  constexpr int BM = 128, BN = 128, BKM = 64;
  const int tiles = (n / BM) * (k2 / BN);
  int slices = 1;
  while (tiles * slices < 512 && slices < 64 &&
         (long)slices * BKM * 4 < m)
    slices *= 2;
  HIP_CHECK(hipMemsetAsync(c, 0, (long)n * k2 * sizeof(float), s));
  if (dbias) HIP_CHECK(hipMemsetAsync(dbias, 0, n * sizeof(float), s));
  gemm_a16_glds_tn_kernel<BM, BN, BKM>
      <<<dim3(tiles * slices), dim3(512), 0, s>>>(a, b, c, dbias, m, n, k2,
                                                  slices);
  const int m_full = m - m % BKM;
  if (m_full < m) {
    gemm_tail_tn_kernel<<<dim3(n), dim3(256), 0, s>>>(
        a + (long)m_full * n, b + (long)m_full * k2, c, dbias, m - m_full, n,
        k2);
  }
}

// ---------------------------------------------------------------------------
// Skinny NN dgrad: dx[m,k2] = g[m,n] . w[n,k2] for TINY m (the P-table
// backward: rpctype has ~5 rows).  The tiled kernel's boundary path runs
// per-element guarded scalar staging on 8 underoccupied workgroups
// (measured 85 us for 2.6 MFLOP at [5,512]x[512,512]); here one wave owns
// one (row, 64-column block) output strip — w reads are lane-coalesced,
// g[i,n] is a broadcast scalar, fully deterministic.
// ---------------------------------------------------------------------------
template <typename OT = float>
__global__ void gemm_skinny_nn_kernel(const __bf16* __restrict__ g,
                                      const float* __restrict__ w,
                                      OT* __restrict__ dx, int m, int n,
                                      int k2) {
  const int i = blockIdx.x;           // output row
  const int j0 = blockIdx.y * PERTGNN_WAVE;
  const int j = j0 + threadIdx.x;     // output column
  if (i >= m || j >= k2) return;
  // 8-deep unroll with split accumulators: the w loads of a batch issue
  // together (a bare loop serializes one memory latency per iteration —
  // the whole kernel is a ~20-wave latency chain)
  float a0 = 0.f, a1 = 0.f;
  int t = 0;
  for (; t + 8 <= n; t += 8) {
    float gv[8], wv[8];
#pragma unroll
    for (int u = 0; u < 8; ++u) gv[u] = (float)g[(long)i * n + t + u];
#pragma unroll
    for (int u = 0; u < 8; ++u) wv[u] = w[(long)(t + u) * k2 + j];
#pragma unroll
    for (int u = 0; u < 8; u += 2) {
      a0 += gv[u] * wv[u];
      a1 += gv[u + 1] * wv[u + 1];
    }
  }
  for (; t < n; ++t) a0 += (float)g[(long)i * n + t] * w[(long)t * k2 + j];
  dx[(long)i * k2 + j] = (OT)(a0 + a1);
}

void launch_gemm_skinny_nn(const void* g_v, const float* w, float* dx, int m,
                           int n, int k2, hipStream_t s) {
  gemm_skinny_nn_kernel<<<dim3(m, (k2 + PERTGNN_WAVE - 1) / PERTGNN_WAVE),
                          dim3(PERTGNN_WAVE), 0, s>>>(
      (const __bf16*)g_v, w, dx, m, n, k2);
}
