// Shared helpers for the pertgnn CDNA4 (gfx950) kernel library.
// Wave size is 64 on CDNA — every cross-lane idiom below is 64-wide.
#pragma once

#include <hip/hip_runtime.h>

#include <stdexcept>
#include <string>

#define PERTGNN_WAVE 64

#define HIP_CHECK(expr)                                                        \
  do {                                                                         \
    hipError_t _e = (expr);                                                    \
    if (_e != hipSuccess) {                                                    \
      throw std::runtime_error(std::string("HIP error ") +                     \
                               hipGetErrorString(_e) + " at " + __FILE__ +     \
                               ":" + std::to_string(__LINE__));                \
    }                                                                          \
  } while (0)

// Launcher shape guard: throws (surfaces as a Python RuntimeError through
// the binding layer) instead of abort()ing the process on a shape the
// kernel template set does not cover.
[[noreturn]] inline void pertgnn_shape_fail(const char* launcher,
                                            const char* what, long v) {
  throw std::runtime_error(std::string("pertgnn ") + launcher +
                           ": unsupported shape (" + what + "=" +
                           std::to_string(v) + ")");
}

// full-wave (64-lane) sum/max reductions via xor shuffles
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}

// reduction over an aligned W-lane sub-group of the wave (W power of two)
template <int W>
__device__ __forceinline__ float subwave_reduce_sum(float v) {
#pragma unroll
  for (int off = W / 2; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, 64));
  return v;
}

static inline int ceil_div(long a, long b) { return (int)((a + b - 1) / b); }

// XCD-aware blockIdx remap (guide §5.5 T1, bijective form): the dispatcher
// places block b on XCD b%8; this remap gives each XCD a CONTIGUOUS chunk of
// the logical grid so neighboring tiles (which share operand panels) hit the
// same XCD-private L2.
__device__ __forceinline__ int xcd_swizzle(int bid, int nwg) {
  const int xcd = bid % 8;
  const int q = nwg / 8;
  const int r = nwg % 8;
  return (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + bid / 8;
}
