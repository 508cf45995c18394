// NHWC cross-channel LRN, fwd + bwd (AlexNet/GoogLeNet norm layers).
// In NHWC the channel window is contiguous memory, so each thread walks a
// sliding window with a running sum — one pass, no re-reads.
// scale = k + alpha/n * sum_win x^2 ;  y = x * scale^-beta
// dx = dy*scale^-beta - (2*alpha*beta/n) * x * sum_win(dy*y/scale)
// scale is stored fp32 for backward (matches the CPU reference exactly).

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

// one thread per pixel: running-window over C
__global__ void lrn_fwd_kernel(const u16* __restrict__ x,
                               u16* __restrict__ y,
                               float* __restrict__ scale,
                               int64_t npix, int C, int half, float a_over_n,
                               float beta, float k) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < npix; i += (int64_t)gridDim.x * blockDim.x) {
    const u16* xp = x + i * C;
    u16* yp = y + i * C;
    float* sp = scale + i * C;
    float win = 0.f;
    for (int c = 0; c <= half && c < C; ++c) {
      float v = ld_bf(xp + c);
      win += v * v;
    }
    for (int c = 0; c < C; ++c) {
      float sc = k + a_over_n * win;
      sp[c] = sc;
      st_bf(yp + c, ld_bf(xp + c) * __powf(sc, -beta));
      int add = c + half + 1, sub = c - half;
      if (add < C) { float v = ld_bf(xp + add); win += v * v; }
      if (sub >= 0) { float v = ld_bf(xp + sub); win -= v * v; }
    }
  }
}

__global__ void lrn_bwd_kernel(const u16* __restrict__ x,
                               const u16* __restrict__ y,
                               const float* __restrict__ scale,
                               const u16* __restrict__ dy,
                               u16* __restrict__ dx,
                               int64_t npix, int C, int half, float a_over_n,
                               float beta, float ratio_coef) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < npix; i += (int64_t)gridDim.x * blockDim.x) {
    const u16* xp = x + i * C;
    const u16* yp = y + i * C;
    const u16* dyp = dy + i * C;
    const float* sp = scale + i * C;
    u16* dxp = dx + i * C;
    auto ratio = [&](int c) {
      return ld_bf(dyp + c) * ld_bf(yp + c) / sp[c];
    };
    float win = 0.f;
    for (int c = 0; c <= half && c < C; ++c) win += ratio(c);
    for (int c = 0; c < C; ++c) {
      float v = ld_bf(dyp + c) * __powf(sp[c], -beta)
                - ratio_coef * ld_bf(xp + c) * win;
      st_bf(dxp + c, v);
      int add = c + half + 1, sub = c - half;
      if (add < C) win += ratio(add);
      if (sub >= 0) win -= ratio(sub);
    }
  }
}

void lrn_fwd(const void* x, void* y, float* scale, int64_t npix, int C,
             int local_size, float alpha, float beta, float k,
             hipStream_t stream) {
  int blocks = (int)hmin<int64_t>(8192, (npix + 255) / 256);
  lrn_fwd_kernel<<<blocks, 256, 0, stream>>>(
      (const u16*)x, (u16*)y, scale, npix, C, local_size / 2,
      alpha / local_size, beta, k);
}

void lrn_bwd(const void* x, const void* y, const float* scale, const void* dy,
             void* dx, int64_t npix, int C, int local_size, float alpha,
             float beta, hipStream_t stream) {
  int blocks = (int)hmin<int64_t>(8192, (npix + 255) / 256);
  lrn_bwd_kernel<<<blocks, 256, 0, stream>>>(
      (const u16*)x, (const u16*)y, scale, (const u16*)dy, (u16*)dx,
      npix, C, local_size / 2, alpha / local_size, beta,
      2.f * alpha * beta / local_size);
}

}  // namespace cosamd
