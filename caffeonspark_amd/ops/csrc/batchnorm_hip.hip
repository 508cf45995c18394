#include "hip/hip_runtime.h"
// NHWC BatchNorm (caffe semantics: normalization only — affine is the
// separate Scale layer).  Named in the north-star hot-path kernel list
// (BASELINE.json; reference engine: upstream BVLC batch_norm_layer.cu).
//
// Train fwd: per-channel mean/biased-var over (N,H,W) via an
// adaptive-block two-moment reduce (same shape as colsum: bx col-groups
// of 8 bf16 channels x by row-groups), then an 8-wide normalize pass.
// Backward: two more per-channel reduces (sum dy, sum dy*xhat), then
//   dx = invstd * (dy - s1/m - xhat * s2/m)
// Global-stats mode reuses the same kernels with s1 = s2 = 0.

#include "common.h"

namespace cosamd {

typedef unsigned short u16;
typedef unsigned short u16x8 __attribute__((ext_vector_type(8)));

__device__ __forceinline__ float ldbf(const u16* p) {
  return bf2f(*reinterpret_cast<const bf16*>(p));
}
__device__ __forceinline__ void stbf(u16* p, float v) {
  bf16 b = f2bf(v);
  *p = *reinterpret_cast<u16*>(&b);
}

// two-moment reduce: out0[c] += sum(w0), out1[c] += sum(w1)
// MODE 0: a = x:     w0 = x,  w1 = x*x      (forward stats)
// MODE 1: a = dy, x_aux = xhat (the BN output — caffe BN has no affine,
//         so y IS xhat even for in-place layers): w0 = dy, w1 = dy*xhat
template <int MODE>
__global__ void bn_reduce_kernel(const u16* __restrict__ a,
                                 const u16* __restrict__ x_aux,
                                 float* __restrict__ out0,
                                 float* __restrict__ out1,
                                 int64_t rows, int C, int bx) {
  __shared__ float part[4096];           // by * bx * 8 * 2 <= 256*8*2
  int by = blockDim.x / bx;
  int c8 = threadIdx.x % bx;
  int rg = threadIdx.x / bx;
  for (int c0 = 0; c0 < C; c0 += bx * 8) {
    int c = c0 + c8 * 8;
    float a0[8] = {}, a1[8] = {};
    if (c + 8 <= C) {
      for (int64_t r = (int64_t)blockIdx.x * by + rg; r < rows;
           r += (int64_t)gridDim.x * by) {
        u16x8 v = *reinterpret_cast<const u16x8*>(a + r * C + c);
        u16x8 w;
        if (MODE == 1)
          w = *reinterpret_cast<const u16x8*>(x_aux + r * C + c);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float f = ldbf((const u16*)&v + j);
          a0[j] += f;
          if (MODE == 0) {
            a1[j] += f * f;
          } else {
            a1[j] += f * ldbf((const u16*)&w + j);
          }
        }
      }
    } else if (c < C) {
      for (int64_t r = (int64_t)blockIdx.x * by + rg; r < rows;
           r += (int64_t)gridDim.x * by)
        for (int j = 0; c + j < C; ++j) {
          float f = ldbf(a + r * C + c + j);
          a0[j] += f;
          if (MODE == 0) {
            a1[j] += f * f;
          } else {
            a1[j] += f * ldbf(x_aux + r * C + c + j);
          }
        }
    }
    int pass_cols = bx * 8;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      part[rg * pass_cols + c8 * 8 + j] = a0[j];
      part[2048 + rg * pass_cols + c8 * 8 + j] = a1[j];
    }
    __syncthreads();
    for (int col = threadIdx.x; col < pass_cols; col += blockDim.x) {
      if (c0 + col < C) {
        float s0 = 0.f, s1 = 0.f;
        for (int g = 0; g < by; ++g) {
          s0 += part[g * pass_cols + col];
          s1 += part[2048 + g * pass_cols + col];
        }
        atomicAdd(out0 + c0 + col, s0);
        atomicAdd(out1 + c0 + col, s1);
      }
    }
    __syncthreads();
  }
}

// y = (x - mean[c]) * invstd[c], 8-wide along channels
__global__ void bn_norm_kernel(const u16* __restrict__ x,
                               u16* __restrict__ y,
                               const float* __restrict__ mean,
                               const float* __restrict__ invstd,
                               int64_t rows, int C) {
  int64_t total8 = rows * (C / 8);
  int nc8 = C / 8;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < total8; i += (int64_t)gridDim.x * blockDim.x) {
    int64_t r = i / nc8;
    int c = (int)(i % nc8) * 8;
    u16x8 v = *reinterpret_cast<const u16x8*>(x + r * C + c);
    u16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      stbf((u16*)&o + j,
           (ldbf((const u16*)&v + j) - mean[c + j]) * invstd[c + j]);
    *reinterpret_cast<u16x8*>(y + r * C + c) = o;
  }
}

// dx = invstd * (dy - s1/m - xhat * s2/m); pass s1 = s2 = zeros and
// inv_m = 0 for the global-stats mode (dx = dy * invstd)
__global__ void bn_bwd_kernel(const u16* __restrict__ xhat,
                              const u16* __restrict__ dy,
                              u16* __restrict__ dx,
                              const float* __restrict__ invstd,
                              const float* __restrict__ s1,
                              const float* __restrict__ s2,
                              float inv_m, int64_t rows, int C) {
  int64_t total8 = rows * (C / 8);
  int nc8 = C / 8;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < total8; i += (int64_t)gridDim.x * blockDim.x) {
    int64_t r = i / nc8;
    int c = (int)(i % nc8) * 8;
    u16x8 vx = *reinterpret_cast<const u16x8*>(xhat + r * C + c);
    u16x8 vd = *reinterpret_cast<const u16x8*>(dy + r * C + c);
    u16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float d = ldbf((const u16*)&vd + j);
      stbf((u16*)&o + j,
           invstd[c + j] * (d - s1[c + j] * inv_m -
                            ldbf((const u16*)&vx + j) * s2[c + j] * inv_m));
    }
    *reinterpret_cast<u16x8*>(dx + r * C + c) = o;
  }
}

// ------------------------------------------------------------- host side

static void bn_reduce_launch(int mode, const void* a, const void* x_aux,
                             float* out0, float* out1, int64_t rows, int C,
                             hipStream_t stream) {
  int bx = hmin<int>(32, (C + 7) / 8);
  int nthreads = 256 - (256 % bx);
  int by = nthreads / bx;
  int64_t work = (rows * C) / (256 * 8) + 1;
  int blocks = (int)hmin<int64_t>(hmin<int64_t>((rows + by - 1) / by, 2048),
                                  work);
  if (mode == 0)
   hipLaunchKernelGGL(( bn_reduce_kernel<0>), dim3(blocks), dim3(nthreads), 0, stream, 
        (const u16*)a, nullptr, out0, out1, rows, C, bx);
  else
   hipLaunchKernelGGL(( bn_reduce_kernel<1>), dim3(blocks), dim3(nthreads), 0, stream, 
        (const u16*)a, (const u16*)x_aux, out0, out1, rows, C, bx);
}

void bn_stats(const void* x, float* sum, float* sumsq, int64_t rows, int C,
              hipStream_t stream) {
  bn_reduce_launch(0, x, nullptr, sum, sumsq, rows, C, stream);
}

void bn_bwd_sums(const void* dy, const void* xhat, float* s1, float* s2,
                 int64_t rows, int C, hipStream_t stream) {
  bn_reduce_launch(1, dy, xhat, s1, s2, rows, C, stream);
}

void bn_norm(const void* x, void* y, const float* mean, const float* invstd,
             int64_t rows, int C, hipStream_t stream) {
  int64_t total8 = rows * (C / 8);
  int blocks = (int)hmin<int64_t>(4096, (total8 + 255) / 256);
 hipLaunchKernelGGL(( bn_norm_kernel), dim3(blocks), dim3(256), 0, stream, (const u16*)x, (u16*)y, mean,
                                             invstd, rows, C);
}

void bn_bwd(const void* xhat, const void* dy, void* dx,
            const float* invstd, const float* s1, const float* s2,
            float inv_m, int64_t rows, int C, hipStream_t stream) {
  int64_t total8 = rows * (C / 8);
  int blocks = (int)hmin<int64_t>(4096, (total8 + 255) / 256);
 hipLaunchKernelGGL(( bn_bwd_kernel), dim3(blocks), dim3(256), 0, stream, 
      (const u16*)xhat, (const u16*)dy, (u16*)dx, invstd, s1, s2, inv_m,
      rows, C);
}

}  // namespace cosamd
