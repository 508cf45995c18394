// NHWC pooling (MAX with argmax, AVE with caffe divisor semantics).
// Caffe semantics (SURVEY.md §3.6 "Pooling"): ceil-mode output dims; AVE
// divides by the window clipped at the *padded* boundary; MAX backward
// routes dy through the recorded argmax (gather formulation, no atomics).

#include "common.h"

namespace cosamd {

typedef unsigned short u16;
typedef unsigned short u16x8 __attribute__((ext_vector_type(8)));
typedef signed char i8x8 __attribute__((ext_vector_type(8)));

// 8-wide channel-vectorized variants (C % 8 == 0): coalesced 16-byte
// window reads/writes; argmax stored window-relative in int8.
__global__ void maxpool_fwd8_kernel(
    const u16* __restrict__ x, u16* __restrict__ y,
    signed char* __restrict__ idx,
    int N, int H, int W, int C, int P, int Q,
    int kh, int kw, int sh, int sw, int ph, int pw, int64_t total8) {
  int c8s = C / 8;
  for (int64_t i8 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i8 < total8; i8 += (int64_t)gridDim.x * blockDim.x) {
    unsigned i = (unsigned)i8;
    int cc = (i % c8s) * 8;
    unsigned iq = i / c8s;
    int q = iq % Q;
    unsigned ip = iq / Q;
    int p = ip % P;
    int n = ip / P;
    int h0 = p * sh - ph, w0 = q * sw - pw;
    float best[8];
    int brs[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) { best[j] = -3.4e38f; brs[j] = 0; }
    for (int r = 0; r < kh; ++r) {
      int h = h0 + r;
      if (h < 0 || h >= H) continue;
      for (int s = 0; s < kw; ++s) {
        int w = w0 + s;
        if (w < 0 || w >= W) continue;
        u16x8 v = *reinterpret_cast<const u16x8*>(
            x + (((int64_t)n * H + h) * W + w) * C + cc);
        int rs = r * kw + s;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          u16 raw = v[j];
          float f = bf2f(*reinterpret_cast<const bf16*>(&raw));
          if (f > best[j]) { best[j] = f; brs[j] = rs; }
        }
      }
    }
    u16x8 out;
    i8x8 oidx;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      bf16 b = f2bf(best[j]);
      out[j] = *reinterpret_cast<u16*>(&b);
      oidx[j] = (signed char)brs[j];
    }
    int64_t o = ((((int64_t)n * P + p) * Q + q) * C + cc);
    *reinterpret_cast<u16x8*>(y + o) = out;
    *reinterpret_cast<i8x8*>(idx + o) = oidx;
  }
}

__global__ void maxpool_bwd8_kernel(
    const u16* __restrict__ dy, const signed char* __restrict__ idx,
    u16* __restrict__ dx,
    int N, int H, int W, int C, int P, int Q,
    int kh, int kw, int sh, int sw, int ph, int pw, int64_t total8) {
  int c8s = C / 8;
  for (int64_t i8 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i8 < total8; i8 += (int64_t)gridDim.x * blockDim.x) {
    unsigned i = (unsigned)i8;
    int cc = (i % c8s) * 8;
    unsigned iw = i / c8s;
    int w = iw % W;
    unsigned ih = iw / W;
    int h = ih % H;
    int n = ih / H;
    float acc[8] = {};
    int plo = max(0, (h + ph - kh + sh) / sh), phi = min(P - 1, (h + ph) / sh);
    int qlo = max(0, (w + pw - kw + sw) / sw), qhi = min(Q - 1, (w + pw) / sw);
    for (int p = plo; p <= phi; ++p) {
      int rr = h - (p * sh - ph);
      for (int q = qlo; q <= qhi; ++q) {
        int rs = rr * kw + (w - (q * sw - pw));
        int64_t o = (((int64_t)n * P + p) * Q + q) * C + cc;
        i8x8 iv = *reinterpret_cast<const i8x8*>(idx + o);
        u16x8 dv = *reinterpret_cast<const u16x8*>(dy + o);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          if ((int)iv[j] == rs) {
            u16 raw = dv[j];
            acc[j] += bf2f(*reinterpret_cast<const bf16*>(&raw));
          }
        }
      }
    }
    u16x8 out;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      bf16 b = f2bf(acc[j]);
      out[j] = *reinterpret_cast<u16*>(&b);
    }
    *reinterpret_cast<u16x8*>(
        dx + (((int64_t)n * H + h) * W + w) * C + cc) = out;
  }
}

template <typename IDX>
__global__ void maxpool_fwd_kernel(
    const u16* __restrict__ x, u16* __restrict__ y, IDX* __restrict__ idx,
    int N, int H, int W, int C, int P, int Q,
    int kh, int kw, int sh, int sw, int ph, int pw, int64_t total) {
  for (int64_t i8 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i8 < total; i8 += (int64_t)gridDim.x * blockDim.x) {
    unsigned i = (unsigned)i8;
    int c = i % C;
    unsigned iq = i / C;
    int q = iq % Q;
    unsigned ip = iq / Q;
    int p = ip % P;
    int n = ip / P;
    int h0 = p * sh - ph, w0 = q * sw - pw;
    float best = -3.4e38f;
    int best_rs = 0;  // window-relative argmax (fits int8 for k<=11)
    for (int r = 0; r < kh; ++r) {
      int h = h0 + r;
      if (h < 0 || h >= H) continue;
      for (int s = 0; s < kw; ++s) {
        int w = w0 + s;
        if (w < 0 || w >= W) continue;
        u16 v = x[(((int64_t)n * H + h) * W + w) * C + c];
        float f = bf2f(*reinterpret_cast<const bf16*>(&v));
        if (f > best) { best = f; best_rs = r * kw + s; }
      }
    }
    bf16 out = f2bf(best);
    y[i8] = *reinterpret_cast<u16*>(&out);
    idx[i8] = (IDX)best_rs;
  }
}

template <typename IDX>
__global__ void maxpool_bwd_kernel(
    const u16* __restrict__ dy, const IDX* __restrict__ idx,
    u16* __restrict__ dx,
    int N, int H, int W, int C, int P, int Q,
    int kh, int kw, int sh, int sw, int ph, int pw, int64_t total) {
  for (int64_t i8 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i8 < total; i8 += (int64_t)gridDim.x * blockDim.x) {
    unsigned i = (unsigned)i8;
    int c = i % C;
    unsigned iw = i / C;
    int w = iw % W;
    unsigned ih = iw / W;
    int h = ih % H;
    int n = ih / H;
    float acc = 0.f;
    // windows (p,q) that can contain (h,w); match window-relative argmax
    int plo = max(0, (h + ph - kh + sh) / sh), phi = min(P - 1, (h + ph) / sh);
    int qlo = max(0, (w + pw - kw + sw) / sw), qhi = min(Q - 1, (w + pw) / sw);
    for (int p = plo; p <= phi; ++p) {
      int rr = h - (p * sh - ph);
      for (int q = qlo; q <= qhi; ++q) {
        int64_t o = (((int64_t)n * P + p) * Q + q) * C + c;
        if ((int)idx[o] == rr * kw + (w - (q * sw - pw))) {
          u16 v = dy[o];
          acc += bf2f(*reinterpret_cast<const bf16*>(&v));
        }
      }
    }
    bf16 out = f2bf(acc);
    dx[i8] = *reinterpret_cast<u16*>(&out);
  }
}

__global__ void avgpool_fwd8_kernel(
    const u16* __restrict__ x, u16* __restrict__ y,
    int N, int H, int W, int C, int P, int Q,
    int kh, int kw, int sh, int sw, int ph, int pw, int64_t total8) {
  int c8s = C / 8;
  for (int64_t i8 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i8 < total8; i8 += (int64_t)gridDim.x * blockDim.x) {
    unsigned i = (unsigned)i8;
    int cc = (i % c8s) * 8;
    unsigned iq = i / c8s;
    int q = iq % Q;
    unsigned ip = iq / Q;
    int p = ip % P;
    int n = ip / P;
    int h0 = p * sh - ph, w0 = q * sw - pw;
    int hend = min(h0 + kh, H + ph), wend = min(w0 + kw, W + pw);
    float inv = 1.f / ((hend - h0) * (wend - w0));
    float acc[8] = {};
    for (int h = max(h0, 0); h < min(hend, H); ++h)
      for (int w = max(w0, 0); w < min(wend, W); ++w) {
        u16x8 v = *reinterpret_cast<const u16x8*>(
            x + (((int64_t)n * H + h) * W + w) * C + cc);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          u16 raw = v[j];
          acc[j] += bf2f(*reinterpret_cast<const bf16*>(&raw));
        }
      }
    u16x8 out;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      bf16 b = f2bf(acc[j] * inv);
      out[j] = *reinterpret_cast<u16*>(&b);
    }
    *reinterpret_cast<u16x8*>(
        y + ((((int64_t)n * P + p) * Q + q) * C + cc)) = out;
  }
}

__global__ void avgpool_bwd8_kernel(
    const u16* __restrict__ dy, u16* __restrict__ dx,
    int N, int H, int W, int C, int P, int Q,
    int kh, int kw, int sh, int sw, int ph, int pw, int64_t total8) {
  int c8s = C / 8;
  for (int64_t i8 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i8 < total8; i8 += (int64_t)gridDim.x * blockDim.x) {
    unsigned i = (unsigned)i8;
    int cc = (i % c8s) * 8;
    unsigned iw = i / c8s;
    int w = iw % W;
    unsigned ih = iw / W;
    int h = ih % H;
    int n = ih / H;
    float acc[8] = {};
    int plo = max(0, (h + ph - kh + sh) / sh), phi = min(P - 1, (h + ph) / sh);
    int qlo = max(0, (w + pw - kw + sw) / sw), qhi = min(Q - 1, (w + pw) / sw);
    for (int p = plo; p <= phi; ++p) {
      int h0 = p * sh - ph;
      int hend = min(h0 + kh, H + ph);
      for (int q = qlo; q <= qhi; ++q) {
        int w0 = q * sw - pw;
        int wend = min(w0 + kw, W + pw);
        float inv = 1.f / ((hend - h0) * (wend - w0));
        u16x8 v = *reinterpret_cast<const u16x8*>(
            dy + ((((int64_t)n * P + p) * Q + q) * C + cc));
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          u16 raw = v[j];
          acc[j] += bf2f(*reinterpret_cast<const bf16*>(&raw)) * inv;
        }
      }
    }
    u16x8 out;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      bf16 b = f2bf(acc[j]);
      out[j] = *reinterpret_cast<u16*>(&b);
    }
    *reinterpret_cast<u16x8*>(
        dx + (((int64_t)n * H + h) * W + w) * C + cc) = out;
  }
}

__global__ void avgpool_fwd_kernel(
    const u16* __restrict__ x, u16* __restrict__ y,
    int N, int H, int W, int C, int P, int Q,
    int kh, int kw, int sh, int sw, int ph, int pw, int64_t total) {
  for (int64_t i8 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i8 < total; i8 += (int64_t)gridDim.x * blockDim.x) {
    unsigned i = (unsigned)i8;
    int c = i % C;
    unsigned iq = i / C;
    int q = iq % Q;
    unsigned ip = iq / Q;
    int p = ip % P;
    int n = ip / P;
    int h0 = p * sh - ph, w0 = q * sw - pw;
    // caffe: divisor = window clipped at the padded boundary
    int hend = min(h0 + kh, H + ph), wend = min(w0 + kw, W + pw);
    int pool_size = (hend - h0) * (wend - w0);
    float acc = 0.f;
    for (int h = max(h0, 0); h < min(hend, H); ++h)
      for (int w = max(w0, 0); w < min(wend, W); ++w) {
        u16 v = x[(((int64_t)n * H + h) * W + w) * C + c];
        acc += bf2f(*reinterpret_cast<const bf16*>(&v));
      }
    bf16 out = f2bf(acc / pool_size);
    y[i8] = *reinterpret_cast<u16*>(&out);
  }
}

__global__ void avgpool_bwd_kernel(
    const u16* __restrict__ dy, u16* __restrict__ dx,
    int N, int H, int W, int C, int P, int Q,
    int kh, int kw, int sh, int sw, int ph, int pw, int64_t total) {
  for (int64_t i8 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i8 < total; i8 += (int64_t)gridDim.x * blockDim.x) {
    unsigned i = (unsigned)i8;
    int c = i % C;
    unsigned iw = i / C;
    int w = iw % W;
    unsigned ih = iw / W;
    int h = ih % H;
    int n = ih / H;
    float acc = 0.f;
    int plo = max(0, (h + ph - kh + sh) / sh), phi = min(P - 1, (h + ph) / sh);
    int qlo = max(0, (w + pw - kw + sw) / sw), qhi = min(Q - 1, (w + pw) / sw);
    for (int p = plo; p <= phi; ++p) {
      int h0 = p * sh - ph;
      int hend = min(h0 + kh, H + ph);
      for (int q = qlo; q <= qhi; ++q) {
        int w0 = q * sw - pw;
        int wend = min(w0 + kw, W + pw);
        int pool_size = (hend - h0) * (wend - w0);
        int64_t o = (((int64_t)n * P + p) * Q + q) * C + c;
        u16 v = dy[o];
        acc += bf2f(*reinterpret_cast<const bf16*>(&v)) / pool_size;
      }
    }
    bf16 out = f2bf(acc);
    dx[i8] = *reinterpret_cast<u16*>(&out);
  }
}

static int nblocks_for(int64_t total) {
  return (int)hmin<int64_t>(8192, (total + 255) / 256);
}

void maxpool_fwd(const void* x, void* y, void* idx, bool idx16, int N,
                 int H, int W, int C, int P, int Q, int kh, int kw, int sh,
                 int sw, int ph, int pw, hipStream_t stream) {
  int64_t total = (int64_t)N * P * Q * C;
  if (idx16 && C % 8 == 0) {
    int64_t total8 = total / 8;
    maxpool_fwd8_kernel<<<nblocks_for(total8), 256, 0, stream>>>(
        (const u16*)x, (u16*)y, (signed char*)idx, N, H, W, C, P, Q, kh,
        kw, sh, sw, ph, pw, total8);
  } else if (idx16)
    maxpool_fwd_kernel<signed char><<<nblocks_for(total), 256, 0, stream>>>(
        (const u16*)x, (u16*)y, (signed char*)idx, N, H, W, C, P, Q, kh, kw, sh,
        sw, ph, pw, total);
  else
    maxpool_fwd_kernel<int><<<nblocks_for(total), 256, 0, stream>>>(
        (const u16*)x, (u16*)y, (int*)idx, N, H, W, C, P, Q, kh, kw, sh,
        sw, ph, pw, total);
}

void maxpool_bwd(const void* dy, const void* idx, bool idx16, void* dx,
                 int N, int H, int W, int C, int P, int Q, int kh, int kw,
                 int sh, int sw, int ph, int pw, hipStream_t stream) {
  int64_t total = (int64_t)N * H * W * C;
  if (idx16 && C % 8 == 0) {
    int64_t total8 = total / 8;
    maxpool_bwd8_kernel<<<nblocks_for(total8), 256, 0, stream>>>(
        (const u16*)dy, (const signed char*)idx, (u16*)dx, N, H, W, C, P,
        Q, kh, kw, sh, sw, ph, pw, total8);
  } else if (idx16)
    maxpool_bwd_kernel<signed char><<<nblocks_for(total), 256, 0, stream>>>(
        (const u16*)dy, (const signed char*)idx, (u16*)dx, N, H, W, C, P, Q, kh,
        kw, sh, sw, ph, pw, total);
  else
    maxpool_bwd_kernel<int><<<nblocks_for(total), 256, 0, stream>>>(
        (const u16*)dy, (const int*)idx, (u16*)dx, N, H, W, C, P, Q, kh,
        kw, sh, sw, ph, pw, total);
}

void avgpool_fwd(const void* x, void* y, int N, int H, int W, int C,
                 int P, int Q, int kh, int kw, int sh, int sw, int ph, int pw,
                 hipStream_t stream) {
  int64_t total = (int64_t)N * P * Q * C;
  if (C % 8 == 0) {
    avgpool_fwd8_kernel<<<nblocks_for(total / 8), 256, 0, stream>>>(
        (const u16*)x, (u16*)y, N, H, W, C, P, Q, kh, kw, sh, sw, ph, pw,
        total / 8);
    return;
  }
  avgpool_fwd_kernel<<<nblocks_for(total), 256, 0, stream>>>(
      (const u16*)x, (u16*)y, N, H, W, C, P, Q, kh, kw, sh, sw, ph, pw,
      total);
}

void avgpool_bwd(const void* dy, void* dx, int N, int H, int W, int C,
                 int P, int Q, int kh, int kw, int sh, int sw, int ph, int pw,
                 hipStream_t stream) {
  int64_t total = (int64_t)N * H * W * C;
  if (C % 8 == 0) {
    avgpool_bwd8_kernel<<<nblocks_for(total / 8), 256, 0, stream>>>(
        (const u16*)dy, (u16*)dx, N, H, W, C, P, Q, kh, kw, sh, sw, ph, pw,
        total / 8);
    return;
  }
  avgpool_bwd_kernel<<<nblocks_for(total), 256, 0, stream>>>(
      (const u16*)dy, (u16*)dx, N, H, W, C, P, Q, kh, kw, sh, sw, ph, pw,
      total);
}

}  // namespace cosamd
