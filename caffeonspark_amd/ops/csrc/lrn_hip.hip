#include "hip/hip_runtime.h"
// NHWC cross-channel LRN, fwd + bwd (AlexNet/GoogLeNet norm layers).
// NHWC puts the channel window in contiguous memory, so one thread per
// ELEMENT re-reads its small window with fully coalesced accesses (adjacent
// lanes -> adjacent channels); the redundant window reads hit L1/L2.
// scale = k + alpha/n * sum_win x^2 ;  y = x * scale^-beta
// dx = dy*scale^-beta - (2*alpha*beta/n) * x * sum_win(dy*y/scale)
// Backward is two passes: ratio = dy*y/scale materialized bf16, then the
// windowed sum — 3x less traffic than recomputing ratio per window tap.

#include "common.h"

namespace cosamd {

typedef unsigned short u16;

__device__ __forceinline__ float ld_bf(const u16* p) {
  return bf2f(*reinterpret_cast<const bf16*>(p));
}
__device__ __forceinline__ void st_bf(u16* p, float v) {
  bf16 b = f2bf(v);
  *p = *reinterpret_cast<u16*>(&b);
}

__global__ void lrn_fwd_kernel(const u16* __restrict__ x,
                               u16* __restrict__ y,
                               float* __restrict__ scale,
                               int64_t total, int C, int half, float a_over_n,
                               float beta, float k) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < total; i += (int64_t)gridDim.x * blockDim.x) {
    int c = (int)(i % C);
    int64_t base = i - c;
    int lo = c - half > 0 ? c - half : 0;
    int hi = c + half < C - 1 ? c + half : C - 1;
    float win = 0.f;
    for (int cc = lo; cc <= hi; ++cc) {
      float v = ld_bf(x + base + cc);
      win += v * v;
    }
    float sc = k + a_over_n * win;
    scale[i] = sc;
    st_bf(y + i, ld_bf(x + i) * __powf(sc, -beta));
  }
}

__global__ void lrn_ratio_kernel(const u16* __restrict__ y,
                                 const float* __restrict__ scale,
                                 const u16* __restrict__ dy,
                                 u16* __restrict__ ratio, int64_t total) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < total; i += (int64_t)gridDim.x * blockDim.x)
    st_bf(ratio + i, ld_bf(dy + i) * ld_bf(y + i) / scale[i]);
}

__global__ void lrn_bwd_kernel(const u16* __restrict__ x,
                               const float* __restrict__ scale,
                               const u16* __restrict__ dy,
                               const u16* __restrict__ ratio,
                               u16* __restrict__ dx,
                               int64_t total, int C, int half,
                               float beta, float ratio_coef) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < total; i += (int64_t)gridDim.x * blockDim.x) {
    int c = (int)(i % C);
    int64_t base = i - c;
    int lo = c - half > 0 ? c - half : 0;
    int hi = c + half < C - 1 ? c + half : C - 1;
    float win = 0.f;
    for (int cc = lo; cc <= hi; ++cc) win += ld_bf(ratio + base + cc);
    float v = ld_bf(dy + i) * __powf(scale[i], -beta)
              - ratio_coef * ld_bf(x + i) * win;
    st_bf(dx + i, v);
  }
}

void lrn_fwd(const void* x, void* y, float* scale, int64_t npix, int C,
             int local_size, float alpha, float beta, float k,
             hipStream_t stream) {
  int64_t total = npix * C;
  int blocks = (int)hmin<int64_t>(4096, (total + 255) / 256);
 hipLaunchKernelGGL(( lrn_fwd_kernel), dim3(blocks), dim3(256), 0, stream, 
      (const u16*)x, (u16*)y, scale, total, C, local_size / 2,
      alpha / local_size, beta, k);
}

void lrn_bwd(const void* x, const void* y, const float* scale, const void* dy,
             void* dx, void* ratio, int64_t npix, int C, int local_size,
             float alpha, float beta, hipStream_t stream) {
  int64_t total = npix * C;
  int blocks = (int)hmin<int64_t>(4096, (total + 255) / 256);
 hipLaunchKernelGGL(( lrn_ratio_kernel), dim3(blocks), dim3(256), 0, stream, 
      (const u16*)y, scale, (const u16*)dy, (u16*)ratio, total);
 hipLaunchKernelGGL(( lrn_bwd_kernel), dim3(blocks), dim3(256), 0, stream, 
      (const u16*)x, scale, (const u16*)dy, (const u16*)ratio, (u16*)dx,
      total, C, local_size / 2, beta, 2.f * alpha * beta / local_size);
}

}  // namespace cosamd
