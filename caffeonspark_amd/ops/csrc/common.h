// Shared helpers for the gfx950 (CDNA4) kernel library.
// Wave size is 64 on CDNA4; all kernels hard-code that.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <cstdint>

#define COS_CHECK_HIP(expr)                                                   \
  do {                                                                        \
    hipError_t _e = (expr);                                                   \
    if (_e != hipSuccess) {                                                   \
      throw std::runtime_error(std::string("HIP error: ") +                   \
                               hipGetErrorString(_e) + " at " __FILE__ ":" +  \
                               std::to_string(__LINE__));                     \
    }                                                                         \
  } while (0)

namespace cosamd {

constexpr int kWave = 64;

static inline int ceil_div(int64_t a, int64_t b) {
  return static_cast<int>((a + b - 1) / b);
}

using bf16 = __hip_bfloat16;

__device__ __forceinline__ float bf2f(bf16 v) {
  return __bfloat162float(v);
}
__device__ __forceinline__ bf16 f2bf(float v) {
  return __float2bfloat16(v);
}

// bijective XCD-aware blockIdx swizzle (8 XCDs on MI355X): contiguous
// chunks of the grid land on one XCD so neighboring tiles share L2.
__device__ __forceinline__ int xcd_swizzle(int bid, int nwg) {
  constexpr int kXCD = 8;
  if (nwg < 2 * kXCD) return bid;
  int q = nwg / kXCD, r = nwg % kXCD;
  int xcd = bid % kXCD, pos = bid / kXCD;
  int base = (xcd < r) ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q;
  return base + pos;
}


// host-side min/max (avoid relying on device overloads in host code)
template <typename T> static inline T hmin(T a, T b) { return a < b ? a : b; }
template <typename T> static inline T hmax(T a, T b) { return a > b ? a : b; }

}  // namespace cosamd
