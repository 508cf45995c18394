// Elementwise kernel family: ReLU fwd/bwd, dropout (counter-based RNG),
// fused SGD-momentum update, LSTM unit fwd/bwd, embed gather/scatter,
// column-sum reduction (bias grads), softmax+NLL loss.
// All bf16 traffic is vectorized 8-wide (G13: scalar bf16 ~2x slower).

#include "common.h"

namespace cosamd {

typedef unsigned short u16;
typedef unsigned short u16x8 __attribute__((ext_vector_type(8)));
typedef unsigned short u16x4 __attribute__((ext_vector_type(4)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

__device__ __forceinline__ float ldbf(const u16* p) {
  return bf2f(*reinterpret_cast<const bf16*>(p));
}
__device__ __forceinline__ void stbf(u16* p, float v) {
  bf16 b = f2bf(v);
  *p = *reinterpret_cast<u16*>(&b);
}

// ---------------------------------------------------------------- ReLU

__global__ void relu_fwd_kernel(const u16* __restrict__ x, u16* __restrict__ y,
                                float slope, int64_t n8, int64_t n) {
  if (blockIdx.x == 0 && threadIdx.x == 0)
    for (int64_t k = n8 * 8; k < n; ++k) {
      float f = ldbf(x + k);
      stbf(y + k, f > 0.f ? f : f * slope);
    }
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < n8; i += (int64_t)gridDim.x * blockDim.x) {
    u16x8 v = *reinterpret_cast<const u16x8*>(x + i * 8);
    u16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      u16 raw = v[j];
      float f = bf2f(*reinterpret_cast<const bf16*>(&raw));
      f = f > 0.f ? f : f * slope;
      bf16 b = f2bf(f);
      o[j] = *reinterpret_cast<u16*>(&b);
    }
    *reinterpret_cast<u16x8*>(y + i * 8) = o;
  }
}

__global__ void relu_bwd_kernel(const u16* __restrict__ y,
                                const u16* __restrict__ dy,
                                u16* __restrict__ dx, float slope,
                                int64_t n8, int64_t n) {
  if (blockIdx.x == 0 && threadIdx.x == 0)
    for (int64_t k = n8 * 8; k < n; ++k) {
      float fy = ldbf(y + k), fd = ldbf(dy + k);
      stbf(dx + k, fy > 0.f ? fd : fd * slope);
    }
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < n8; i += (int64_t)gridDim.x * blockDim.x) {
    u16x8 vy = *reinterpret_cast<const u16x8*>(y + i * 8);
    u16x8 vd = *reinterpret_cast<const u16x8*>(dy + i * 8);
    u16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      u16 ry = vy[j], rd = vd[j];
      float fy = bf2f(*reinterpret_cast<const bf16*>(&ry));
      float fd = bf2f(*reinterpret_cast<const bf16*>(&rd));
      float g = fy > 0.f ? fd : fd * slope;
      bf16 b = f2bf(g);
      o[j] = *reinterpret_cast<u16*>(&b);
    }
    *reinterpret_cast<u16x8*>(dx + i * 8) = o;
  }
}

// relu backward over a row-strided channel slice (fused-concat branch:
// y and dy are channel windows of channels_last buffers, row stride =
// the concat's total channel count); dx written dense [rows][C]
__global__ void relu_bwd_strided_kernel(const u16* __restrict__ y,
                                        const u16* __restrict__ dy,
                                        u16* __restrict__ dx, float slope,
                                        int64_t rows, int C, int ldy,
                                        int lddy) {
  int nc8 = C / 8;
  bool vec = (C % 8 == 0);
  int64_t total = rows * (vec ? nc8 : C);
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < total; i += (int64_t)gridDim.x * blockDim.x) {
    if (vec) {
      int64_t r = i / nc8;
      int c = (int)(i % nc8) * 8;
      u16x8 vy = *reinterpret_cast<const u16x8*>(y + r * ldy + c);
      u16x8 vd = *reinterpret_cast<const u16x8*>(dy + r * lddy + c);
      u16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        u16 ry = vy[j], rd = vd[j];
        float fy = bf2f(*reinterpret_cast<const bf16*>(&ry));
        float fd = bf2f(*reinterpret_cast<const bf16*>(&rd));
        float g = fy > 0.f ? fd : fd * slope;
        bf16 b = f2bf(g);
        o[j] = *reinterpret_cast<u16*>(&b);
      }
      *reinterpret_cast<u16x8*>(dx + r * C + c) = o;
    } else {
      int64_t r = i / C;
      int c = (int)(i % C);
      float fy = ldbf(y + r * ldy + c);
      float fd = ldbf(dy + r * lddy + c);
      stbf(dx + r * C + c, fy > 0.f ? fd : fd * slope);
    }
  }
}

void relu_bwd_strided(const void* y, const void* dy, void* dx, float slope,
                      int64_t rows, int C, int ldy, int lddy,
                      hipStream_t stream) {
  int64_t total = rows * ((C % 8 == 0) ? C / 8 : C);
  int blocks = (int)hmin<int64_t>(4096, (total + 255) / 256);
  relu_bwd_strided_kernel<<<blocks, 256, 0, stream>>>(
      (const u16*)y, (const u16*)dy, (u16*)dx, slope, rows, C, ldy, lddy);
}

// -------------------------------------------------------------- dropout
// counter-based RNG (splitmix-style hash of seed+index): reproducible,
// stateless, good enough for dropout masks.

__device__ __forceinline__ unsigned hash_rng(uint64_t seed, uint64_t idx) {
  uint64_t z = seed + idx * 0x9E3779B97F4A7C15ull;
  z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ull;
  z = (z ^ (z >> 27)) * 0x94D049BB133111EBull;
  return (unsigned)(z >> 32);
}

// seed lives in device memory (advanced by seed_bump below) so a step
// captured in a hipGraph draws a fresh mask on every replay
__global__ void seed_bump_kernel(unsigned long long* s) {
  *s = *s * 6364136223846793005ull + 1442695040888963407ull;
}

__global__ void dropout_fwd_kernel(const u16* __restrict__ x,
                                   u16* __restrict__ y, u16* __restrict__ mask,
                                   float keep, float inv_keep,
                                   const unsigned long long* __restrict__ seedp,
                                   int64_t n) {
  uint64_t seed = *seedp;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < n; i += (int64_t)gridDim.x * blockDim.x) {
    float u = (hash_rng(seed, i) >> 8) * (1.f / 16777216.f);
    float m = (u < keep) ? inv_keep : 0.f;
    stbf(mask + i, m);
    stbf(y + i, ldbf(x + i) * m);
  }
}

__global__ void mul_bf16_kernel(const u16* __restrict__ a,
                                const u16* __restrict__ b,
                                u16* __restrict__ y, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < n; i += (int64_t)gridDim.x * blockDim.x)
    stbf(y + i, ldbf(a + i) * ldbf(b + i));
}

// ------------------------------------------------------- SGD fused update
// V = mu*V + lr*(g + wd*p); p -= V   (fp32 master weights)

__global__ void sgd_update_kernel(float* __restrict__ p, const float* __restrict__ g,
                                  float* __restrict__ v, float lr, float mu,
                                  float wd, int64_t n4, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < n4; i += (int64_t)gridDim.x * blockDim.x) {
    if (i * 4 + 4 <= n) {
      f32x4 pv = *reinterpret_cast<f32x4*>(p + i * 4);
      f32x4 gv = *reinterpret_cast<const f32x4*>(g + i * 4);
      f32x4 vv = *reinterpret_cast<f32x4*>(v + i * 4);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        float gg = gv[j] + wd * pv[j];
        vv[j] = mu * vv[j] + lr * gg;
        pv[j] -= vv[j];
      }
      *reinterpret_cast<f32x4*>(v + i * 4) = vv;
      *reinterpret_cast<f32x4*>(p + i * 4) = pv;
    } else {
      for (int64_t k = i * 4; k < n; ++k) {
        float gg = g[k] + wd * p[k];
        v[k] = mu * v[k] + lr * gg;
        p[k] -= v[k];
      }
    }
  }
}

// ------------------------------------------------------- bias+act+cast
// Finalize pass for split-K forward GEMMs: bf16(relu(acc + bias[col])).

__global__ void bias_act_cast_kernel(const float* __restrict__ in,
                                     const float* __restrict__ bias,
                                     u16* __restrict__ out, int64_t rows,
                                     int cols, int relu) {
  int64_t total = rows * cols;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < total; i += (int64_t)gridDim.x * blockDim.x) {
    float v = in[i];
    if (bias != nullptr) v += bias[i % cols];
    if (relu && v < 0.f) v = 0.f;
    stbf(out + i, v);
  }
}

void bias_act_cast(const float* in, const float* bias, void* out,
                   int64_t rows, int cols, bool relu, hipStream_t stream) {
  int64_t total = rows * cols;
  int blocks = (int)hmin<int64_t>(4096, (total + 255) / 256);
  bias_act_cast_kernel<<<blocks, 256, 0, stream>>>(
      in, bias, (u16*)out, rows, cols, relu ? 1 : 0);
}

// ------------------------------------------------------------ transpose
// Tiled 2D bf16 transpose [R][C] -> [C][R]: 64x64 tiles staged through
// LDS (row pad 66 u16 -> conflict-free transposed reads), coalesced
// 128-byte accesses on both sides.  Used to turn K-major backward GEMM
// operands into the fast NT direct form.

__global__ void transpose_bf16_kernel(const u16* __restrict__ in,
                                      u16* __restrict__ out,
                                      int R, int C, int tiles_c) {
  // 128(rows) x 64(cols) tiles; row-PAIRS packed as u32 in LDS (halves
  // LDS op count — b16 LDS traffic runs at half rate); write phase maps
  // 16 threads per output row so each store instruction covers 256
  // contiguous bytes; XCD-swizzled tile order for L2 row reuse.
  __shared__ unsigned lds32[64][67];
  int nblk = gridDim.x;
  int tile = xcd_swizzle(blockIdx.x, nblk);
  int r0 = (tile / tiles_c) * 128;
  int c0 = (tile % tiles_c) * 64;
  int t = threadIdx.x;
  bool interior = (r0 + 128 <= R) && (c0 + 64 <= C);
  int lc8 = (t & 7) * 8;               // 8-wide column slot
#pragma unroll
  for (int half = 0; half < 2; ++half) {
    int r2 = (t >> 3) + half * 32;     // row pair 0..63
    int grlo = r0 + 2 * r2, grhi = grlo + 1;
    if (interior) {
      u16x8 vlo = *reinterpret_cast<const u16x8*>(
          in + (int64_t)grlo * C + c0 + lc8);
      u16x8 vhi = *reinterpret_cast<const u16x8*>(
          in + (int64_t)grhi * C + c0 + lc8);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        lds32[r2][lc8 + j] = (unsigned)vlo[j] | ((unsigned)vhi[j] << 16);
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        int gc = c0 + lc8 + j;
        u16 lo = (grlo < R && gc < C) ? in[(int64_t)grlo * C + gc] : 0;
        u16 hi = (grhi < R && gc < C) ? in[(int64_t)grhi * C + gc] : 0;
        lds32[r2][lc8 + j] = (unsigned)lo | ((unsigned)hi << 16);
      }
    }
  }
  __syncthreads();
  int cl = t >> 4;                      // 16 output rows (orig cols)/pass
  int lr8 = (t & 15) * 8;               // 16 threads x 8 = 128 orig rows
#pragma unroll
  for (int pass = 0; pass < 4; ++pass) {
    int c = cl + pass * 16;
    int gc = c0 + c;
    if (interior) {
      u16x8 v;
#pragma unroll
      for (int m = 0; m < 4; ++m) {
        unsigned pair = lds32[(lr8 >> 1) + m][c];
        v[2 * m] = (u16)(pair & 0xFFFF);
        v[2 * m + 1] = (u16)(pair >> 16);
      }
      *reinterpret_cast<u16x8*>(out + (int64_t)gc * R + r0 + lr8) = v;
    } else if (gc < C) {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        int gr = r0 + lr8 + j;
        unsigned pair = lds32[(lr8 + j) >> 1][c];
        u16 v = ((lr8 + j) & 1) ? (u16)(pair >> 16) : (u16)(pair & 0xFFFF);
        if (gr < R) out[(int64_t)gc * R + gr] = v;
      }
    }
  }
}

void transpose_bf16(const void* in, void* out, int64_t R, int64_t C,
                    hipStream_t stream) {
  int tiles_r = (int)((R + 127) / 128), tiles_c = (int)((C + 63) / 64);
  transpose_bf16_kernel<<<tiles_r * tiles_c, 256, 0, stream>>>(
      (const u16*)in, (u16*)out, (int)R, (int)C, tiles_c);
}

// one launch for ALL (lr_mult, decay_mult) segments of the flat arena:
// blocks are striped across segments proportionally to their size
// segment table cached in LDS (a per-quad global scan serializes on
// dependent loads: measured 241us -> bandwidth-bound after this)
#define COS_SGD_MAX_SEG 512
__global__ void sgd_update_multi_kernel(
    float* __restrict__ p, const float* __restrict__ g,
    float* __restrict__ v, const int64_t* __restrict__ seg_off,
    const float* __restrict__ seg_lr, const float* __restrict__ seg_wd,
    int nseg, float mu, int64_t total4, u16* __restrict__ sh) {
  __shared__ int64_t soff[COS_SGD_MAX_SEG + 1];
  __shared__ float slr[COS_SGD_MAX_SEG], swd[COS_SGD_MAX_SEG];
  for (int i = threadIdx.x; i <= nseg; i += blockDim.x) {
    soff[i] = seg_off[i];
    if (i < nseg) { slr[i] = seg_lr[i]; swd[i] = seg_wd[i]; }
  }
  __syncthreads();
  int s = 0;  // a thread's quads are monotonically increasing, so the
              // segment cursor only moves forward: amortized O(nseg)
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < total4; i += (int64_t)gridDim.x * blockDim.x) {
    int64_t e0 = i * 4;
    while (s + 1 < nseg && e0 >= soff[s + 1]) ++s;
    float lr = slr[s], wd = swd[s];
    if (lr == 0.f) continue;
    if (soff[s + 1] >= e0 + 4) {         // interior quad: vector path
      f32x4 pv = *reinterpret_cast<f32x4*>(p + e0);
      f32x4 gv = *reinterpret_cast<const f32x4*>(g + e0);
      f32x4 vv = *reinterpret_cast<f32x4*>(v + e0);
#pragma unroll
      for (int k = 0; k < 4; ++k) {
        float gg = gv[k] + wd * pv[k];
        vv[k] = mu * vv[k] + lr * gg;
        pv[k] -= vv[k];
      }
      *reinterpret_cast<f32x4*>(v + e0) = vv;
      *reinterpret_cast<f32x4*>(p + e0) = pv;
      if (sh != nullptr) {
        // bf16 shadow write fused in: the separate whole-arena cast
        // pass (and its read of p) disappears from the step
        u16x4 o;
#pragma unroll
        for (int k = 0; k < 4; ++k) {
          bf16 b = f2bf(pv[k]);
          o[k] = *reinterpret_cast<u16*>(&b);
        }
        *reinterpret_cast<u16x4*>(sh + e0) = o;
      }
    } else {
      int64_t end = soff[s + 1];
      for (int64_t k = e0; k < end; ++k) {
        float gg = g[k] + wd * p[k];
        v[k] = mu * v[k] + lr * gg;
        p[k] -= v[k];
        if (sh != nullptr) {
          bf16 b = f2bf(p[k]);
          sh[k] = *reinterpret_cast<u16*>(&b);
        }
      }
    }
  }
}

void sgd_update_multi(float* p, const float* g, float* v,
                      const int64_t* seg_off, const float* seg_lr,
                      const float* seg_wd, int nseg, float mu,
                      int64_t total, void* sh, hipStream_t stream) {
  int64_t total4 = (total + 3) / 4;
  int blocks = (int)hmin<int64_t>(2048, (total4 + 255) / 256);
  sgd_update_multi_kernel<<<blocks, 256, 0, stream>>>(
      p, g, v, seg_off, seg_lr, seg_wd, nseg, mu, total4, (u16*)sh);
}

// Nesterov momentum, same whole-arena structure (reference solver
// family, SURVEY.md §2.5: caffe Nesterov p -= (1+mu)*v_new - mu*v_old)
__global__ void nesterov_update_multi_kernel(
    float* __restrict__ p, const float* __restrict__ g,
    float* __restrict__ v, const int64_t* __restrict__ seg_off,
    const float* __restrict__ seg_lr, const float* __restrict__ seg_wd,
    int nseg, float mu, int64_t total4, u16* __restrict__ sh) {
  __shared__ int64_t soff[COS_SGD_MAX_SEG + 1];
  __shared__ float slr[COS_SGD_MAX_SEG], swd[COS_SGD_MAX_SEG];
  for (int i = threadIdx.x; i <= nseg; i += blockDim.x) {
    soff[i] = seg_off[i];
    if (i < nseg) { slr[i] = seg_lr[i]; swd[i] = seg_wd[i]; }
  }
  __syncthreads();
  int s = 0;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < total4; i += (int64_t)gridDim.x * blockDim.x) {
    int64_t e0 = i * 4;
    while (s + 1 < nseg && e0 >= soff[s + 1]) ++s;
    float lr = slr[s], wd = swd[s];
    if (lr == 0.f) continue;
    int64_t end = min(soff[s + 1], e0 + 4);
    for (int64_t k = e0; k < end; ++k) {
      float gg = g[k] + wd * p[k];
      float vprev = v[k];
      float vnew = mu * vprev + lr * gg;
      v[k] = vnew;
      p[k] -= (1.f + mu) * vnew - mu * vprev;
      if (sh != nullptr) {
        bf16 b = f2bf(p[k]);
        sh[k] = *reinterpret_cast<u16*>(&b);
      }
    }
  }
}

void nesterov_update_multi(float* p, const float* g, float* v,
                           const int64_t* seg_off, const float* seg_lr,
                           const float* seg_wd, int nseg, float mu,
                           int64_t total, void* sh, hipStream_t stream) {
  int64_t total4 = (total + 3) / 4;
  int blocks = (int)hmin<int64_t>(2048, (total4 + 255) / 256);
  nesterov_update_multi_kernel<<<blocks, 256, 0, stream>>>(
      p, g, v, seg_off, seg_lr, seg_wd, nseg, mu, total4, (u16*)sh);
}

// Adam, whole-arena: m/v moment arenas; the bias-correction factor is
// folded into seg_lr host-side (it is global per step)
__global__ void adam_update_multi_kernel(
    float* __restrict__ p, const float* __restrict__ g,
    float* __restrict__ m, float* __restrict__ v,
    const int64_t* __restrict__ seg_off,
    const float* __restrict__ seg_lr, const float* __restrict__ seg_wd,
    int nseg, float b1, float b2, float eps, int64_t total4,
    u16* __restrict__ sh) {
  __shared__ int64_t soff[COS_SGD_MAX_SEG + 1];
  __shared__ float slr[COS_SGD_MAX_SEG], swd[COS_SGD_MAX_SEG];
  for (int i = threadIdx.x; i <= nseg; i += blockDim.x) {
    soff[i] = seg_off[i];
    if (i < nseg) { slr[i] = seg_lr[i]; swd[i] = seg_wd[i]; }
  }
  __syncthreads();
  int s = 0;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < total4; i += (int64_t)gridDim.x * blockDim.x) {
    int64_t e0 = i * 4;
    while (s + 1 < nseg && e0 >= soff[s + 1]) ++s;
    float lr = slr[s], wd = swd[s];
    if (lr == 0.f) continue;
    int64_t end = min(soff[s + 1], e0 + 4);
    for (int64_t k = e0; k < end; ++k) {
      float gg = g[k] + wd * p[k];
      float mm = b1 * m[k] + (1.f - b1) * gg;
      float vv = b2 * v[k] + (1.f - b2) * gg * gg;
      m[k] = mm;
      v[k] = vv;
      p[k] -= lr * mm / (sqrtf(vv) + eps);
      if (sh != nullptr) {
        bf16 b = f2bf(p[k]);
        sh[k] = *reinterpret_cast<u16*>(&b);
      }
    }
  }
}

void adam_update_multi(float* p, const float* g, float* m, float* v,
                       const int64_t* seg_off, const float* seg_lr,
                       const float* seg_wd, int nseg, float b1, float b2,
                       float eps, int64_t total, void* sh,
                       hipStream_t stream) {
  int64_t total4 = (total + 3) / 4;
  int blocks = (int)hmin<int64_t>(2048, (total4 + 255) / 256);
  adam_update_multi_kernel<<<blocks, 256, 0, stream>>>(
      p, g, m, v, seg_off, seg_lr, seg_wd, nseg, b1, b2, eps, total4,
      (u16*)sh);
}

// ---------------------------------------------------------- column reduce
// out[c] += sum_r in[r*ld + c]  (bias gradients; in bf16, out fp32)
// Block = 256 threads as [4 row-groups x 64 cols]: coalesced 64-wide column
// reads, deep grid.x row parallelism, LDS-reduced partials, one atomic per
// (block, col).

// fused ReLU backward + bias-gradient column sum: dx = relu'(y) * dy
// AND db[c] += column sums of dx, in ONE pass over dy/y (they were two
// separate full passes: relu_bwd_strided + colsum ~10% of the GoogLeNet
// step combined).  Same adaptive [bx col-groups x by row-groups] shape
// and LDS tree reduce as colsum_kernel; strided y/dy support the
// fused-concat channel-window views.
__global__ void relu_colsum_bwd_kernel(
    const u16* __restrict__ y, const u16* __restrict__ dy,
    u16* __restrict__ dx, float* __restrict__ db, float slope,
    int64_t rows, int cols, int ldy, int lddy, int bx) {
  __shared__ float part[2048];
  int by = blockDim.x / bx;
  int c8 = threadIdx.x % bx;
  int rg = threadIdx.x / bx;
  for (int c0 = 0; c0 < cols; c0 += bx * 8) {
    int c = c0 + c8 * 8;
    float acc[8] = {};
    if (rg < by) {
      if (c + 8 <= cols) {
        for (int64_t r = (int64_t)blockIdx.x * by + rg; r < rows;
             r += (int64_t)gridDim.x * by) {
          u16x8 vy = *reinterpret_cast<const u16x8*>(y + r * ldy + c);
          u16x8 vd = *reinterpret_cast<const u16x8*>(dy + r * lddy + c);
          u16x8 o;
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            float fy = ldbf((const u16*)&vy + j);
            float fd = ldbf((const u16*)&vd + j);
            float g = fy > 0.f ? fd : fd * slope;
            acc[j] += g;
            bf16 b = f2bf(g);
            o[j] = *reinterpret_cast<u16*>(&b);
          }
          *reinterpret_cast<u16x8*>(dx + r * cols + c) = o;
        }
      } else if (c < cols) {
        for (int64_t r = (int64_t)blockIdx.x * by + rg; r < rows;
             r += (int64_t)gridDim.x * by)
          for (int j = 0; c + j < cols; ++j) {
            float fy = ldbf(y + r * ldy + c + j);
            float fd = ldbf(dy + r * lddy + c + j);
            float g = fy > 0.f ? fd : fd * slope;
            acc[j] += g;
            stbf(dx + r * cols + c + j, g);
          }
      }
#pragma unroll
      for (int j = 0; j < 8; ++j) part[rg * bx * 8 + c8 * 8 + j] = acc[j];
    }
    __syncthreads();
    int pass_cols = bx * 8;
    for (int col = threadIdx.x; col < pass_cols; col += blockDim.x) {
      if (c0 + col < cols) {
        float sum = 0.f;
        for (int g = 0; g < by; ++g) sum += part[g * pass_cols + col];
        atomicAdd(db + c0 + col, sum);
      }
    }
    __syncthreads();
  }
}

// 8-wide column reduce with adaptive block shape: bx col-groups (8
// bf16 columns = one 16-byte load each) x by row-groups, bx sized to
// the matrix so narrow matrices (conv bias grads, cols 96-384) still
// keep every lane loading 16 bytes.  LDS tree reduce, one atomicAdd
// per column per block.
__global__ void colsum_kernel(const u16* __restrict__ in, float* __restrict__ out,
                              int64_t rows, int cols, int ld, int bx) {
  __shared__ float part[2048];           // by * bx * 8 <= 256*8
  int by = blockDim.x / bx;
  int c8 = threadIdx.x % bx;
  int rg = threadIdx.x / bx;             // 0..by-1
  for (int c0 = 0; c0 < cols; c0 += bx * 8) {
    int c = c0 + c8 * 8;
    float acc[8] = {};
    if (rg < by) {
      if (c + 8 <= cols) {
        for (int64_t r = (int64_t)blockIdx.x * by + rg; r < rows;
             r += (int64_t)gridDim.x * by) {
          u16x8 v = *reinterpret_cast<const u16x8*>(in + r * ld + c);
#pragma unroll
          for (int j = 0; j < 8; ++j) acc[j] += ldbf((const u16*)&v + j);
        }
      } else if (c < cols) {
        for (int64_t r = (int64_t)blockIdx.x * by + rg; r < rows;
             r += (int64_t)gridDim.x * by)
          for (int j = 0; c + j < cols; ++j)
            acc[j] += ldbf(in + r * ld + c + j);
      }
#pragma unroll
      for (int j = 0; j < 8; ++j) part[rg * bx * 8 + c8 * 8 + j] = acc[j];
    }
    __syncthreads();
    int pass_cols = bx * 8;
    for (int col = threadIdx.x; col < pass_cols; col += blockDim.x) {
      if (c0 + col < cols) {
        float sum = 0.f;
        for (int g = 0; g < by; ++g) sum += part[g * pass_cols + col];
        atomicAdd(out + c0 + col, sum);
      }
    }
    __syncthreads();
  }
}

// --------------------------------------------------------------- LSTM unit
// gates [N,4H] pre-activation (i,f,o,g); fp32 cell state.

__global__ void lstm_unit_fwd_kernel(
    const float* __restrict__ c_prev, const u16* __restrict__ gates,
    const u16* __restrict__ cont, float* __restrict__ c_out,
    u16* __restrict__ h_out, float* __restrict__ act,  // [N,4,H] activated
    int64_t n, int H) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < n; i += (int64_t)gridDim.x * blockDim.x) {
    int64_t row = i / H;
    int hh = i % H;
    const u16* g = gates + row * 4 * H;
    float gi = 1.f / (1.f + __expf(-ldbf(g + hh)));
    float gf = 1.f / (1.f + __expf(-ldbf(g + H + hh)));
    float go = 1.f / (1.f + __expf(-ldbf(g + 2 * H + hh)));
    float gg = tanhf(ldbf(g + 3 * H + hh));
    float ct = ldbf(cont + row);
    float c = gf * c_prev[i] * ct + gi * gg;
    float tc = tanhf(c);
    c_out[i] = c;
    stbf(h_out + i, go * tc);
    float* a = act + row * 4 * H;
    a[hh] = gi; a[H + hh] = gf; a[2 * H + hh] = go; a[3 * H + hh] = gg;
    // tc recomputed in bwd from c_out
  }
}

__global__ void lstm_unit_bwd_kernel(
    const float* __restrict__ c_prev, const float* __restrict__ c_out,
    const float* __restrict__ act, const u16* __restrict__ cont,
    const float* __restrict__ dc_next, const u16* __restrict__ dh,
    float* __restrict__ dc_prev, u16* __restrict__ dgates,
    int64_t n, int H) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < n; i += (int64_t)gridDim.x * blockDim.x) {
    int64_t row = i / H;
    int hh = i % H;
    const float* a = act + row * 4 * H;
    float gi = a[hh], gf = a[H + hh], go = a[2 * H + hh], gg = a[3 * H + hh];
    float tc = tanhf(c_out[i]);
    float ct = ldbf(cont + row);
    float dhv = ldbf(dh + i);
    float dc = dc_next[i] + dhv * go * (1.f - tc * tc);
    float dov = dhv * tc;
    float div = dc * gg;
    float dgv = dc * gi;
    float dfv = dc * c_prev[i] * ct;
    dc_prev[i] = dc * gf * ct;
    u16* dg = dgates + row * 4 * H;
    stbf(dg + hh, div * gi * (1.f - gi));
    stbf(dg + H + hh, dfv * gf * (1.f - gf));
    stbf(dg + 2 * H + hh, dov * go * (1.f - go));
    stbf(dg + 3 * H + hh, dgv * (1.f - gg * gg));
  }
}

// ------------------------------------------------------------------ embed

__global__ void embed_fwd_kernel(const float* __restrict__ idx,
                                 const u16* __restrict__ w,
                                 u16* __restrict__ y, int64_t n, int E,
                                 int V) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < n * E; i += (int64_t)gridDim.x * blockDim.x) {
    int64_t row = i / E;
    int e = i % E;
    int v = (int)idx[row];
    v = v < 0 ? 0 : (v >= V ? V - 1 : v);
    y[i] = w[(int64_t)v * E + e];
  }
}

__global__ void embed_bwd_kernel(const float* __restrict__ idx,
                                 const u16* __restrict__ dy,
                                 float* __restrict__ dw, int64_t n, int E,
                                 int V) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < n * E; i += (int64_t)gridDim.x * blockDim.x) {
    int64_t row = i / E;
    int e = i % E;
    int v = (int)idx[row];
    if (v < 0 || v >= V) continue;
    atomicAdd(dw + (int64_t)v * E + e, ldbf(dy + i));
  }
}

// ----------------------------------------------------- softmax + NLL loss
// x [N, C] bf16 -> prob fp32, per-row loss accumulated into loss[0],
// valid count into count[0].  One wave per row.

__global__ void softmax_loss_fwd_kernel(
    const u16* __restrict__ x, const float* __restrict__ label,
    float* __restrict__ prob, float* __restrict__ loss,
    int* __restrict__ count, int64_t nrows, int C, int ignore,
    int has_ignore) {
  int64_t row = (int64_t)blockIdx.x * (blockDim.x / 64) + (threadIdx.x / 64);
  int lane = threadIdx.x & 63;
  if (row >= nrows) return;
  const u16* xp = x + row * C;
  float mx = -3.4e38f;
  for (int c = lane; c < C; c += 64) mx = fmaxf(mx, ldbf(xp + c));
#pragma unroll
  for (int off = 32; off; off >>= 1)
    mx = fmaxf(mx, __shfl_down(mx, off));
  mx = __shfl(mx, 0);
  float sum = 0.f;
  for (int c = lane; c < C; c += 64) sum += __expf(ldbf(xp + c) - mx);
#pragma unroll
  for (int off = 32; off; off >>= 1) sum += __shfl_down(sum, off);
  sum = __shfl(sum, 0);
  float inv = 1.f / sum;
  float* pp = prob + row * C;
  for (int c = lane; c < C; c += 64)
    pp[c] = __expf(ldbf(xp + c) - mx) * inv;
  if (lane == 0) {
    int lab = (int)label[row];
    if (!(has_ignore && lab == ignore)) {
      float lp = ldbf(xp + lab) - mx - __logf(sum);
      atomicAdd(loss, -lp);
      atomicAdd(count, 1);
    }
  }
}

__global__ void softmax_loss_bwd_kernel(
    const float* __restrict__ prob, const float* __restrict__ label,
    u16* __restrict__ dx, float scale, int64_t nrows, int C, int ignore,
    int has_ignore) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < nrows * C; i += (int64_t)gridDim.x * blockDim.x) {
    int64_t row = i / C;
    int c = i % C;
    int lab = (int)label[row];
    float v;
    if (has_ignore && lab == ignore) {
      v = 0.f;
    } else {
      v = prob[i] - (c == lab ? 1.f : 0.f);
    }
    stbf(dx + i, v * scale);
  }
}

// ------------------------------------------------------------------- hosts

static int nb(int64_t total) {
  return (int)hmin<int64_t>(4096, (total + 255) / 256);
}

void relu_fwd(const void* x, void* y, float slope, int64_t n,
              hipStream_t stream) {
  int64_t n8 = n / 8;
  relu_fwd_kernel<<<nb(hmax<int64_t>(n8, 1)), 256, 0, stream>>>(
      (const u16*)x, (u16*)y, slope, n8, n);
}

void relu_bwd(const void* y, const void* dy, void* dx, float slope, int64_t n,
              hipStream_t stream) {
  int64_t n8 = n / 8;
  relu_bwd_kernel<<<nb(hmax<int64_t>(n8, 1)), 256, 0, stream>>>(
      (const u16*)y, (const u16*)dy, (u16*)dx, slope, n8, n);
}

void seed_bump(void* s, hipStream_t stream) {
  seed_bump_kernel<<<1, 1, 0, stream>>>((unsigned long long*)s);
}

void dropout_fwd(const void* x, void* y, void* mask, float ratio,
                 const void* seed, int64_t n, hipStream_t stream) {
  float keep = 1.f - ratio;
  dropout_fwd_kernel<<<nb(n), 256, 0, stream>>>(
      (const u16*)x, (u16*)y, (u16*)mask, keep, 1.f / keep, (const unsigned long long*)seed, n);
}

void mul_bf16(const void* a, const void* b, void* y, int64_t n,
              hipStream_t stream) {
  mul_bf16_kernel<<<nb(n), 256, 0, stream>>>(
      (const u16*)a, (const u16*)b, (u16*)y, n);
}

void sgd_update(float* p, const float* g, float* v, float lr, float mu,
                float wd, int64_t n, hipStream_t stream) {
  int64_t n4 = (n + 3) / 4;
  sgd_update_kernel<<<nb(n4), 256, 0, stream>>>(p, g, v, lr, mu, wd, n4, n);
}

void relu_colsum_bwd(const void* y, const void* dy, void* dx, float* db,
                     float slope, int64_t rows, int cols, int ldy,
                     int lddy, hipStream_t stream) {
  int bx = hmin<int>(32, (cols + 7) / 8);
  int nthreads = 256 - (256 % bx);
  int by = nthreads / bx;
  // dx writes dominate (unlike colsum): size the grid for throughput
  // but keep the per-column atomic count bounded
  int64_t work_blocks = (rows * hmax<int64_t>(cols, 1)) / (256 * 16) + 1;
  int blocks = (int)hmin<int64_t>(hmin<int64_t>((rows + by - 1) / by, 4096),
                                  work_blocks);
  relu_colsum_bwd_kernel<<<blocks, nthreads, 0, stream>>>(
      (const u16*)y, (const u16*)dy, (u16*)dx, db, slope, rows, cols,
      ldy, lddy, bx);
}

void colsum(const void* in, float* out, int64_t rows, int cols, int ld,
            hipStream_t stream) {
  // scale the grid to the actual element count so small reductions don't
  // dispatch thousands of idle blocks (then atomic-sum their zeros)
  int bx = hmin<int>(32, (cols + 7) / 8);
  int nthreads = 256 - (256 % bx);       // whole row-groups only
  int by = nthreads / bx;
  // every block atomicAdds all `cols` partials, so block count must
  // scale with the element count (>=64 elements per thread) or small
  // reductions drown in atomics (GoogLeNet bias grads: 30us -> ~8us)
  int64_t work_blocks = (rows * hmax<int64_t>(cols, 1)) / (256 * 64) + 1;
  int blocks = (int)hmin<int64_t>(hmin<int64_t>((rows + by - 1) / by, 2048),
                                  work_blocks);
  colsum_kernel<<<blocks, nthreads, 0, stream>>>((const u16*)in, out, rows,
                                                 cols, ld, bx);
}

void lstm_unit_fwd(const float* c_prev, const void* gates, const void* cont,
                   float* c_out, void* h_out, float* act, int64_t n, int H,
                   hipStream_t stream) {
  lstm_unit_fwd_kernel<<<nb(n), 256, 0, stream>>>(
      c_prev, (const u16*)gates, (const u16*)cont, c_out, (u16*)h_out, act,
      n, H);
}

void lstm_unit_bwd(const float* c_prev, const float* c_out, const float* act,
                   const void* cont, const float* dc_next, const void* dh,
                   float* dc_prev, void* dgates, int64_t n, int H,
                   hipStream_t stream) {
  lstm_unit_bwd_kernel<<<nb(n), 256, 0, stream>>>(
      c_prev, c_out, act, (const u16*)cont, dc_next, (const u16*)dh, dc_prev,
      (u16*)dgates, n, H);
}

void embed_fwd(const float* idx, const void* w, void* y, int64_t n, int E,
               int V, hipStream_t stream) {
  embed_fwd_kernel<<<nb(n * E), 256, 0, stream>>>(
      idx, (const u16*)w, (u16*)y, n, E, V);
}

void embed_bwd(const float* idx, const void* dy, float* dw, int64_t n, int E,
               int V, hipStream_t stream) {
  embed_bwd_kernel<<<nb(n * E), 256, 0, stream>>>(
      idx, (const u16*)dy, dw, n, E, V);
}

void softmax_loss_fwd(const void* x, const float* label, float* prob,
                      float* loss, int* count, int64_t nrows, int C,
                      int ignore, bool has_ignore, hipStream_t stream) {
  int waves_per_block = 4;
  int64_t blocks = (nrows + waves_per_block - 1) / waves_per_block;
  unsigned grid = (unsigned)hmin<int64_t>(blocks, (int64_t)1 << 30);
  softmax_loss_fwd_kernel<<<grid, waves_per_block * 64, 0, stream>>>(
      (const u16*)x, label, prob, loss, count, nrows, C, ignore,
      has_ignore ? 1 : 0);
}

void softmax_loss_bwd(const float* prob, const float* label, void* dx,
                      float scale, int64_t nrows, int C, int ignore,
                      bool has_ignore, hipStream_t stream) {
  softmax_loss_bwd_kernel<<<nb(nrows * C), 256, 0, stream>>>(
      prob, label, (u16*)dx, scale, nrows, C, ignore, has_ignore ? 1 : 0);
}

}  // namespace cosamd
