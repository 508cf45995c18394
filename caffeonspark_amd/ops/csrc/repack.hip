// Fused conv-weight repack: ONE launch refreshes every conv layer's
// GEMM-ready weight layouts per step, replacing per-layer permute-copy
// + transpose kernels (GoogLeNet: 57 convs -> ~120 tiny launches/step,
// each dominated by wave-ramp overhead).
//
// Per descriptor: src  [Kout][Cg][R][S] bf16 (the solver's bf16 shadow
//                 view of the caffe-layout master weight)
//                 dst  [>=Kout][Kpad]   (forward GEMM B operand,
//                 [k][(r*S+s)*Cg + c]; pad columns stay zero)
//                 dstT [>=Kpad][Kout]   (optional, 0 = skip: the dx
//                 GEMM's transposed operand)
//                 dstF [G][>=Cg][K2p]   (optional, 0 = skip: flipped
//                 per-group layout [g][c][(RS-1-rs)*Kg + k%Kg] for the
//                 implicit-dx GEMM; pad columns stay zero)
// Table rows: [src, dst, dstT, Kout, Cg, R, S, Kpad, dstF, K2p, G, Cgp]
// (Cgp = the padded row count of one dstF group slab).

#include "common.h"

namespace cosamd {

typedef unsigned short u16;

__global__ void repack_weights_kernel(const int64_t* __restrict__ table,
                                      int ndesc) {
  const int64_t* e = table + (int64_t)blockIdx.y * 12;
  const u16* src = reinterpret_cast<const u16*>(e[0]);
  u16* dst = reinterpret_cast<u16*>(e[1]);
  u16* dstT = reinterpret_cast<u16*>(e[2]);
  int Kout = (int)e[3], Cg = (int)e[4], R = (int)e[5], S = (int)e[6];
  int Kpad = (int)e[7];
  u16* dstF = reinterpret_cast<u16*>(e[8]);
  int K2p = (int)e[9];
  int G = (int)e[10], Cgp = (int)e[11];
  int Kg = G > 0 ? Kout / G : Kout;
  int RS = R * S;
  int Kcol = RS * Cg;
  int64_t total = (int64_t)Kout * Kcol;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < total; i += (int64_t)gridDim.x * blockDim.x) {
    int k = (int)(i / Kcol);
    int col = (int)(i % Kcol);
    int rs = col / Cg, c = col % Cg;
    u16 v = src[((int64_t)k * Cg + c) * RS + rs];
    dst[(int64_t)k * Kpad + col] = v;
    if (dstT != nullptr) dstT[(int64_t)col * Kout + k] = v;
    if (dstF != nullptr) {
      int g = k / Kg;
      dstF[((int64_t)g * Cgp + c) * K2p +
           (RS - 1 - rs) * Kg + (k - g * Kg)] = v;
    }
  }
}

void repack_weights(const int64_t* table, int ndesc, int64_t max_total,
                    hipStream_t stream) {
  if (ndesc <= 0) return;
  int bx = (int)hmin<int64_t>(512, (max_total + 255) / 256);
  dim3 grid(bx, ndesc);
  repack_weights_kernel<<<grid, 256, 0, stream>>>(table, ndesc);
}

// conv weight-gradient unpack + arena accumulate: the dw GEMM produces
// dwp[Kout][Kpad] in the GEMM's [k][(r*S+s)*Cg + c] layout; this writes
// it into the caffe-layout fp32 arena slice [Kout][Cg][R][S] in one
// pass (overwrite or +=), replacing a permute-contiguous copy plus the
// arena accumulate.
__global__ void dw_unpack_acc_kernel(const float* __restrict__ dwp,
                                     float* __restrict__ arena,
                                     int Kout, int Cg, int R, int S,
                                     int Kpad, int accumulate) {
  int RS = R * S;
  int Kcol = RS * Cg;
  int64_t total = (int64_t)Kout * Kcol;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < total; i += (int64_t)gridDim.x * blockDim.x) {
    int k = (int)(i / Kcol);
    int crs = (int)(i % Kcol);   // arena-order within one filter
    int c = crs / RS, rs = crs % RS;
    float v = dwp[(int64_t)k * Kpad + rs * Cg + c];
    if (accumulate)
      arena[i] += v;
    else
      arena[i] = v;
  }
}

void dw_unpack_acc(const float* dwp, float* arena, int Kout, int Cg,
                   int R, int S, int Kpad, bool accumulate,
                   hipStream_t stream) {
  int64_t total = (int64_t)Kout * R * S * Cg;
  int blocks = (int)hmin<int64_t>(2048, (total + 255) / 256);
  dw_unpack_acc_kernel<<<blocks, 256, 0, stream>>>(
      dwp, arena, Kout, Cg, R, S, Kpad, accumulate ? 1 : 0);
}

}  // namespace cosamd
