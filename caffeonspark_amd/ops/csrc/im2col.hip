// NHWC im2col / col2im for the implicit-GEMM convolution path.
//
// col[(n*P+p)*Q+q][(r*S+s)*C + c] = x[n][h][w][c],  h = p*sh - ph + r, etc.
// The K (column) dimension is padded to a multiple of 8 (Kpad) so the GEMM's
// 16-byte direct-to-LDS staging path stays aligned; pad columns are zeroed.
// NHWC makes every (r,s) slice a C-contiguous copy — coalesced on both ends.
// col2im is the gather formulation (no atomics): each input pixel sums the
// col entries that reference it.

#include "common.h"

namespace cosamd {

typedef unsigned short u16;
typedef unsigned short u16x8 __attribute__((ext_vector_type(8)));

// column-major variant: one thread per 8 output columns — fully coalesced
// 16-byte writes AND (for contiguous c-runs) coalesced reads; a per-block
// LDS lookup table kills the per-element div/mod for k -> (dh, dw, c).
__global__ void im2col_smallc_kernel(
    const u16* __restrict__ x, u16* __restrict__ col,
    int N, int H, int W, int C, int P, int Q,
    int R, int S, int sh, int sw, int ph, int pw,
    int dil, int Kpad, int c0, int Ct, int64_t total8) {
  // LUT transposed to [j][kslot] so a wavefront's reads (adjacent
  // threads = adjacent kslots) hit consecutive banks — the natural
  // [k] layout strides by 8 ints and 16-way-conflicts (PMC: 1.39
  // conflicts/busy-cycle before, ~0 after)
  __shared__ int lut[2560];  // (dh<<20)|(dw<<10)|c, or -1 for pad columns
  int Kcol = R * S * Ct;
  for (int k = threadIdx.x; k < Kpad; k += blockDim.x) {
    int e;
    if (k < Kcol) {
      int rs = k / Ct, c = k % Ct;
      e = (((rs / S) * dil) << 20) | (((rs % S) * dil) << 10) | c;
    } else {
      e = -1;
    }
    lut[(k & 7) * 320 + (k >> 3)] = e;
  }
  __syncthreads();
  int kslots = Kpad / 8;
  // Ct%8==0 makes every 8-slot a contiguous channel run of ONE (r,s):
  // one 16-byte gather instead of 8 scalar loads (conv2-5 of AlexNet,
  // all GoogLeNet 3x3/5x5 towers; conv1's Ct=3 takes the scalar path)
  bool runs8 = (Ct % 8 == 0);
  // pad-0 dil-1 full-channel small-C convs (AlexNet conv1): k runs are
  // contiguous (s,c) spans of one input row, always in-bounds — ONE
  // unaligned 16-byte load (gfx950 supports it natively) replaces the
  // 8 scalar gathers; chunks straddling an r boundary merge two.
  int SC = S * C;
  // SC >= 8 keeps a chunk within at most two filter rows (the merge
  // below handles exactly one straddle; LeNet's 5x1 runs would span 3)
  bool runfast = !runs8 && Ct == C && c0 == 0 && dil == 1 &&
                 ph == 0 && pw == 0 && SC >= 8;
  int64_t total_x = (int64_t)N * H * W * C;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < total8; i += (int64_t)gridDim.x * blockDim.x) {
    int64_t npq = i / kslots;
    int k0 = (int)(i % kslots) * 8;
    int q = npq % Q;
    int p = (npq / Q) % P;
    int n = npq / ((int64_t)P * Q);
    int h0 = p * sh - ph, w0 = q * sw - pw;
    const u16* xbase = x + (int64_t)n * H * W * C + c0;
    int ks = k0 >> 3;
    u16x8 out;
    int e = lut[ks];
    if (runs8) {
      if (e >= 0) {
        int h = h0 + (e >> 20);
        int w = w0 + ((e >> 10) & 1023);
        if (h >= 0 && h < H && w >= 0 && w < W) {
          out = *reinterpret_cast<const u16x8*>(
              xbase + ((int64_t)h * W + w) * C + (e & 1023));
        } else {
          out = u16x8{0, 0, 0, 0, 0, 0, 0, 0};
        }
      } else {
        out = u16x8{0, 0, 0, 0, 0, 0, 0, 0};
      }
    } else {
      int k0f = ks * 8;
      const u16* src = nullptr;
      int r0 = 0, t0 = 0, sp = 8;
      if (runfast && k0f + 8 <= Kcol) {
        r0 = k0f / SC;
        t0 = k0f - r0 * SC;
        sp = SC - t0;                 // elements before r increments
        src = xbase + (int64_t)(h0 + r0) * W * C + w0 * C + t0;
        if (src - x + 8 > total_x) src = nullptr;  // over-read guard
      }
      if (src != nullptr && sp >= 8) {
        __builtin_memcpy(&out, src, 16);
      } else if (src != nullptr) {
        u16x8 lo, hi;
        __builtin_memcpy(&lo, src, 16);
        const u16* src2 = xbase + (int64_t)(h0 + r0 + 1) * W * C + w0 * C;
        __builtin_memcpy(&hi, src2, 16);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          out[j] = j < sp ? lo[j] : hi[j - sp];
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          int ej = lut[j * 320 + ks];
          u16 v = 0;
          if (ej >= 0) {
            int h = h0 + (ej >> 20);
            int w = w0 + ((ej >> 10) & 1023);
            if (h >= 0 && h < H && w >= 0 && w < W)
              v = xbase[((int64_t)h * W + w) * C + (ej & 1023)];
          }
          out[j] = v;
        }
      }
    }
    *reinterpret_cast<u16x8*>(col + npq * Kpad + k0) = out;
  }
}

__global__ void im2col_nhwc_kernel(
    const u16* __restrict__ x, u16* __restrict__ col,
    int N, int H, int W, int C, int P, int Q,
    int R, int S, int sh, int sw, int ph, int pw,
    int dil, int Kpad, int c0, int Ct, unsigned total_rs) {
  // one thread per (npq, r, s): adjacent threads share npq and walk rs,
  // so writes are contiguous along the col row; 32-bit index math
  unsigned RS = R * S;
  for (unsigned i = blockIdx.x * blockDim.x + threadIdx.x;
       i < total_rs; i += gridDim.x * blockDim.x) {
    int rs = i % RS;
    unsigned npq = i / RS;
    int q = npq % Q;
    unsigned pq = npq / Q;
    int p = pq % P;
    int n = pq / P;
    int r = rs / S, s = rs % S;
    int h = p * sh - ph + r * dil;
    int w = q * sw - pw + s * dil;
    u16* dst = col + npq * Kpad + rs * Ct;
    if (h >= 0 && h < H && w >= 0 && w < W) {
      const u16* src = x + (((int64_t)n * H + h) * W + w) * C + c0;
      int c = 0;
      for (; c + 8 <= Ct; c += 8)
        *reinterpret_cast<u16x8*>(dst + c) =
            *reinterpret_cast<const u16x8*>(src + c);
      for (; c < Ct; ++c) dst[c] = src[c];
    } else {
      int c = 0;
      for (; c + 8 <= Ct; c += 8)
        *reinterpret_cast<u16x8*>(dst + c) = u16x8{0, 0, 0, 0, 0, 0, 0, 0};
      for (; c < Ct; ++c) dst[c] = 0;
    }
  }
}

__global__ void col_zero_pad_kernel(u16* __restrict__ col, int Kcol, int Kpad,
                                    int64_t rows) {
  int padw = Kpad - Kcol;
  int64_t total = rows * padw;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < total; i += (int64_t)gridDim.x * blockDim.x) {
    int64_t row = i / padw;
    col[row * Kpad + Kcol + (i % padw)] = 0;
  }
}

// stride-1 vectorized col2im: one thread per 8 channels of one input
// pixel — no divisions in the window walk (p = h+ph-r), 16-byte
// coalesced dcol reads and dx writes.
__global__ void col2im_s1_kernel(
    const u16* __restrict__ dcol, u16* __restrict__ dx,
    int N, int H, int W, int C, int P, int Q,
    int R, int S, int ph, int pw, int Kpad, int c0, int Ct,
    int64_t total8) {
  int c8s = Ct / 8;
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
    for (int r = 0; r < R; ++r) {
      int p = h + ph - r;
      if (p < 0 || p >= P) continue;
      for (int s = 0; s < S; ++s) {
        int q = w + pw - s;
        if (q < 0 || q >= Q) continue;
        int64_t npq = ((int64_t)n * P + p) * Q + q;
        u16x8 v = *reinterpret_cast<const u16x8*>(
            dcol + npq * Kpad + (r * S + s) * Ct + cc);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          u16 raw = v[j];
          acc[j] += bf2f(*reinterpret_cast<const bf16*>(&raw));
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
        dx + (((int64_t)n * H + h) * W + w) * C + c0 + cc) = out;
  }
}

// dcol -> dx (gather, fp32 accumulate, bf16 out)
__global__ void col2im_nhwc_kernel(
    const u16* __restrict__ dcol, u16* __restrict__ dx,
    int N, int H, int W, int C, int P, int Q,
    int R, int S, int sh, int sw, int ph, int pw,
    int dil, int Kpad, int c0, int Ct, int64_t total) {
  for (int64_t i8 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i8 < total; i8 += (int64_t)gridDim.x * blockDim.x) {
    unsigned i = (unsigned)i8;
    int c = i % Ct;
    unsigned iw = i / Ct;
    int w = iw % W;
    unsigned ih = iw / W;
    int h = ih % H;
    int n = ih / H;
    float acc = 0.f;
    for (int r = 0; r < R; ++r) {
      int hp = h + ph - r * dil;
      if (hp < 0 || hp % sh) continue;
      int p = hp / sh;
      if (p >= P) continue;
      for (int s = 0; s < S; ++s) {
        int wp = w + pw - s * dil;
        if (wp < 0 || wp % sw) continue;
        int q = wp / sw;
        if (q >= Q) continue;
        int64_t npq = ((int64_t)n * P + p) * Q + q;
        u16 v = dcol[npq * Kpad + (r * S + s) * Ct + c];
        acc += bf2f(*reinterpret_cast<const bf16*>(&v));
      }
    }
    bf16 out = f2bf(acc);
    dx[(((int64_t)n * H + h) * W + w) * C + c0 + c] =
        *reinterpret_cast<u16*>(&out);
  }
}

void im2col_nhwc(const void* x, void* col, int N, int H, int W, int C,
                 int P, int Q, int R, int S, int sh, int sw, int ph, int pw,
                 int dil, int Kpad, int c0, int Ct, hipStream_t stream) {
  if (Kpad <= 2560) {
    int64_t total8 = (int64_t)N * P * Q * (Kpad / 8);
    int b = hmin<int64_t>(8192, (total8 + 255) / 256);
    im2col_smallc_kernel<<<b, 256, 0, stream>>>(
        (const u16*)x, (u16*)col, N, H, W, C, P, Q, R, S, sh, sw, ph, pw,
        dil, Kpad, c0, Ct, total8);
    return;  // pad columns are zero-filled inline
  }
  int64_t total = (int64_t)N * P * Q * R * S;
  int blocks = hmin<int64_t>(4096, (total + 255) / 256);
  im2col_nhwc_kernel<<<blocks, 256, 0, stream>>>(
      (const u16*)x, (u16*)col, N, H, W, C, P, Q, R, S, sh, sw, ph, pw,
      dil, Kpad, c0, Ct, (unsigned)total);
  int Kcol = R * S * Ct;
  if (Kpad > Kcol) {
    int64_t rows = (int64_t)N * P * Q;
    int b2 = hmin<int64_t>(2048, (rows * (Kpad - Kcol) + 255) / 256);
    col_zero_pad_kernel<<<b2, 256, 0, stream>>>((u16*)col, Kcol, Kpad, rows);
  }
}

// Transposed im2col for the dW backward GEMM's B operand: writes
// colT[k][npq] (k = (r*S+s)*Cg + c, row stride NPQ) straight from the
// NHWC input.  Replaces an explicit transpose of the forward col matrix:
// the transpose reads + writes the full col matrix (the largest tensor in
// a conv backward), while this reads only the R*S-times-smaller input
// (L2-resident) and does the same coalesced writes.  Block = 32 npq-slots
// x 8 k-rows; k tiles on blockIdx.x so the round-robin XCD dispatch gives
// every XCD the same small input window to reuse.
__global__ void im2col_t_kernel(
    const u16* __restrict__ x, u16* __restrict__ out,
    int N, int H, int W, int C, int P, int Q, int R, int S,
    int sh, int sw, int ph, int pw, int dil, int c0, int Cg,
    int64_t NPQ) {
  int k = blockIdx.x * 8 + threadIdx.y;
  int r = k / (S * Cg), rem = k % (S * Cg);
  int s = rem / Cg, c = c0 + rem % Cg;
  int64_t npq = (int64_t)blockIdx.y * 256 + threadIdx.x * 8;
  if (npq >= NPQ) return;
  int PQ = P * Q;
  int n = (int)(npq / PQ), pq = (int)(npq % PQ);
  int p = pq / Q, q = pq % Q;
  u16x8 v;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    u16 val = 0;
    if (npq + j < NPQ && r < R) {
      int hi = p * sh - ph + r * dil;
      int wi = q * sw - pw + s * dil;
      if (hi >= 0 && hi < H && wi >= 0 && wi < W)
        val = x[(((int64_t)n * H + hi) * W + wi) * C + c];
    }
    v[j] = val;
    if (++q == Q) { q = 0; if (++p == P) { p = 0; ++n; } }
  }
  int64_t o = (int64_t)k * NPQ + npq;
  if (npq + 8 <= NPQ && (o & 7) == 0)
    *reinterpret_cast<u16x8*>(out + o) = v;
  else
    for (int j = 0; j < 8 && npq + j < NPQ; ++j) out[o + j] = v[j];
}

void im2col_t(const void* x, void* colT, int N, int H, int W, int C,
              int P, int Q, int R, int S, int sh, int sw, int ph, int pw,
              int dil, int Kpad, int c0, int Cg, hipStream_t stream) {
  int64_t NPQ = (int64_t)N * P * Q;
  dim3 grid((Kpad + 7) / 8, (unsigned)((NPQ + 255) / 256));
  dim3 block(32, 8);
  im2col_t_kernel<<<grid, block, 0, stream>>>(
      (const u16*)x, (u16*)colT, N, H, W, C, P, Q, R, S, sh, sw, ph, pw,
      dil, c0, Cg, NPQ);
}

void col2im_nhwc(const void* dcol, void* dx, int N, int H, int W, int C,
                 int P, int Q, int R, int S, int sh, int sw, int ph, int pw,
                 int dil, int Kpad, int c0, int Ct, hipStream_t stream) {
  if (sh == 1 && sw == 1 && dil == 1 && Ct % 8 == 0) {
    int64_t total8 = (int64_t)N * H * W * (Ct / 8);
    int b = hmin<int64_t>(8192, (total8 + 255) / 256);
    col2im_s1_kernel<<<b, 256, 0, stream>>>(
        (const u16*)dcol, (u16*)dx, N, H, W, C, P, Q, R, S, ph, pw, Kpad,
        c0, Ct, total8);
    return;
  }
  int64_t total = (int64_t)N * H * W * Ct;
  int blocks = hmin<int64_t>(8192, (total + 255) / 256);
  col2im_nhwc_kernel<<<blocks, 256, 0, stream>>>(
      (const u16*)dcol, (u16*)dx, N, H, W, C, P, Q, R, S, sh, sw, ph, pw,
      dil, Kpad, c0, Ct, total);
}

}  // namespace cosamd
