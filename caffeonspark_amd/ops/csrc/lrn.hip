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
typedef unsigned short u16x8 __attribute__((ext_vector_type(8)));

__device__ __forceinline__ float ld_bf(const u16* p) {
  return bf2f(*reinterpret_cast<const bf16*>(p));
}
__device__ __forceinline__ void st_bf(u16* p, float v) {
  bf16 b = f2bf(v);
  *p = *reinterpret_cast<u16*>(&b);
}

// sc^-beta: __powf costs ~10x a rsqrt, and caffe LRN is almost always
// beta = 0.75 (sc^-3/4 = rsqrt(sc) * sqrt(rsqrt(sc))) or 0.5
template <int BMODE>  // 0: generic powf, 1: beta=0.75, 2: beta=0.5
__device__ __forceinline__ float pow_negbeta(float sc, float beta) {
  if (BMODE == 1) {
    float r = __frsqrt_rn(sc);
    return r * __fsqrt_rn(r);
  }
  if (BMODE == 2) return __frsqrt_rn(sc);
  return __powf(sc, -beta);
}

template <int BMODE>
__global__ void lrn_fwd_kernel(const u16* __restrict__ x,
                               u16* __restrict__ y,
                               float* __restrict__ scale,
                               int64_t total, int C, int half, float a_over_n,
                               float beta, float k) {
  // one thread per 8 channels: three 16-byte octet loads (prev/mid/next)
  // cover the [cc-half, cc+7+half] window (half<=4), float4 scale stores
  int c8s = C / 8;
  for (int64_t i8 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i8 < total; i8 += (int64_t)gridDim.x * blockDim.x) {
    int cc = (int)(i8 % c8s) * 8;
    int64_t base = (i8 / c8s) * (int64_t)C;
    float v[24];
    u16x8 oct;
    if (cc >= 8) {
      oct = *reinterpret_cast<const u16x8*>(x + base + cc - 8);
#pragma unroll
      for (int j = 0; j < 8; ++j) v[j] = ld_bf((const u16*)&oct + j);
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) v[j] = 0.f;
    }
    oct = *reinterpret_cast<const u16x8*>(x + base + cc);
#pragma unroll
    for (int j = 0; j < 8; ++j) v[8 + j] = ld_bf((const u16*)&oct + j);
    if (cc + 16 <= C) {
      oct = *reinterpret_cast<const u16x8*>(x + base + cc + 8);
#pragma unroll
      for (int j = 0; j < 8; ++j) v[16 + j] = ld_bf((const u16*)&oct + j);
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) v[16 + j] = 0.f;
    }
    u16x8 out;
    float scv[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float win = 0.f;
#pragma unroll
      for (int t = 0; t <= 8; ++t)  // window size 2*half+1 <= 9
        if (t <= 2 * half) win += v[8 + j - half + t] * v[8 + j - half + t];
      float sc = k + a_over_n * win;
      scv[j] = sc;
      bf16 b = f2bf(v[8 + j] * pow_negbeta<BMODE>(sc, beta));
      out[j] = *reinterpret_cast<u16*>(&b);
    }
    auto* sp = reinterpret_cast<float4*>(scale + base + cc);
    sp[0] = float4{scv[0], scv[1], scv[2], scv[3]};
    sp[1] = float4{scv[4], scv[5], scv[6], scv[7]};
    *reinterpret_cast<u16x8*>(y + base + cc) = out;
  }
}

__global__ void lrn_ratio_kernel(const u16* __restrict__ y,
                                 const float* __restrict__ scale,
                                 const u16* __restrict__ dy,
                                 u16* __restrict__ ratio, int64_t total8) {
  for (int64_t i8 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i8 < total8; i8 += (int64_t)gridDim.x * blockDim.x) {
    int64_t i = i8 * 8;
    u16x8 vy = *reinterpret_cast<const u16x8*>(y + i);
    u16x8 vd = *reinterpret_cast<const u16x8*>(dy + i);
    auto* sp = reinterpret_cast<const float4*>(scale + i);
    float4 s0 = sp[0], s1 = sp[1];
    u16x8 out;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      u16 ry = vy[j], rd = vd[j];
      float sc = j < 4 ? (&s0.x)[j] : (&s1.x)[j - 4];
      float f = bf2f(*reinterpret_cast<const bf16*>(&rd)) *
                bf2f(*reinterpret_cast<const bf16*>(&ry)) / sc;
      bf16 b = f2bf(f);
      out[j] = *reinterpret_cast<u16*>(&b);
    }
    *reinterpret_cast<u16x8*>(ratio + i) = out;
  }
}

template <int BMODE>
__global__ void lrn_bwd_kernel(const u16* __restrict__ x,
                               const float* __restrict__ scale,
                               const u16* __restrict__ dy,
                               const u16* __restrict__ ratio,
                               u16* __restrict__ dx,
                               int64_t total8, int C, int half,
                               float beta, float ratio_coef) {
  int c8s = C / 8;
  for (int64_t i8 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i8 < total8; i8 += (int64_t)gridDim.x * blockDim.x) {
    int cc = (int)(i8 % c8s) * 8;
    int64_t base = (i8 / c8s) * (int64_t)C;
    float r[24];
    u16x8 oct;
    if (cc >= 8) {
      oct = *reinterpret_cast<const u16x8*>(ratio + base + cc - 8);
#pragma unroll
      for (int j = 0; j < 8; ++j) r[j] = ld_bf((const u16*)&oct + j);
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) r[j] = 0.f;
    }
    oct = *reinterpret_cast<const u16x8*>(ratio + base + cc);
#pragma unroll
    for (int j = 0; j < 8; ++j) r[8 + j] = ld_bf((const u16*)&oct + j);
    if (cc + 16 <= C) {
      oct = *reinterpret_cast<const u16x8*>(ratio + base + cc + 8);
#pragma unroll
      for (int j = 0; j < 8; ++j) r[16 + j] = ld_bf((const u16*)&oct + j);
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) r[16 + j] = 0.f;
    }
    u16x8 vdy = *reinterpret_cast<const u16x8*>(dy + base + cc);
    u16x8 vx = *reinterpret_cast<const u16x8*>(x + base + cc);
    auto* sp = reinterpret_cast<const float4*>(scale + base + cc);
    float4 s0 = sp[0], s1 = sp[1];
    u16x8 out;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float win = 0.f;
#pragma unroll
      for (int t = 0; t <= 8; ++t)
        if (t <= 2 * half) win += r[8 + j - half + t];
      float sc = j < 4 ? (&s0.x)[j] : (&s1.x)[j - 4];
      float v = ld_bf((const u16*)&vdy + j) * pow_negbeta<BMODE>(sc, beta)
                - ratio_coef * ld_bf((const u16*)&vx + j) * win;
      bf16 b = f2bf(v);
      out[j] = *reinterpret_cast<u16*>(&b);
    }
    *reinterpret_cast<u16x8*>(dx + base + cc) = out;
  }
}

void lrn_fwd(const void* x, void* y, float* scale, int64_t npix, int C,
             int local_size, float alpha, float beta, float k,
             hipStream_t stream) {
  // vectorized kernel requires C%8 and window radius <=4 (local_size<=9,
  // true for AlexNet/GoogLeNet's local_size 5); callers with other
  // configs run the CPU-parity fallback in ops/gpu.py
  int64_t total8 = npix * (C / 8);
  int blocks = (int)hmin<int64_t>(4096, (total8 + 255) / 256);
  if (beta == 0.75f)
    lrn_fwd_kernel<1><<<blocks, 256, 0, stream>>>(
        (const u16*)x, (u16*)y, scale, total8, C, local_size / 2,
        alpha / local_size, beta, k);
  else if (beta == 0.5f)
    lrn_fwd_kernel<2><<<blocks, 256, 0, stream>>>(
        (const u16*)x, (u16*)y, scale, total8, C, local_size / 2,
        alpha / local_size, beta, k);
  else
    lrn_fwd_kernel<0><<<blocks, 256, 0, stream>>>(
        (const u16*)x, (u16*)y, scale, total8, C, local_size / 2,
        alpha / local_size, beta, k);
}

void lrn_bwd(const void* x, const void* y, const float* scale, const void* dy,
             void* dx, void* ratio, int64_t npix, int C, int local_size,
             float alpha, float beta, hipStream_t stream) {
  int64_t total8 = npix * (C / 8);
  int blocks = (int)hmin<int64_t>(4096, (total8 + 255) / 256);
  lrn_ratio_kernel<<<blocks, 256, 0, stream>>>(
      (const u16*)y, scale, (const u16*)dy, (u16*)ratio, total8);
  float rc = 2.f * alpha * beta / local_size;
  if (beta == 0.75f)
    lrn_bwd_kernel<1><<<blocks, 256, 0, stream>>>(
        (const u16*)x, scale, (const u16*)dy, (const u16*)ratio, (u16*)dx,
        total8, C, local_size / 2, beta, rc);
  else if (beta == 0.5f)
    lrn_bwd_kernel<2><<<blocks, 256, 0, stream>>>(
        (const u16*)x, scale, (const u16*)dy, (const u16*)ratio, (u16*)dx,
        total8, C, local_size / 2, beta, rc);
  else
    lrn_bwd_kernel<0><<<blocks, 256, 0, stream>>>(
        (const u16*)x, scale, (const u16*)dy, (const u16*)ratio, (u16*)dx,
        total8, C, local_size / 2, beta, rc);
}

}  // namespace cosamd
