// bf16 MFMA GEMM for gfx950 (CDNA4) — the compute core of conv (im2col) and
// inner-product layers, forward and backward.
//
// C[M,N] = A' @ B' with A',B' presented as [rows][K] panels:
//   TRANS_A=false: A global layout [M][K] (lda = row stride)
//   TRANS_A=true : A global layout [K][M] (lda = row stride of the K-major
//                  matrix) — staged transposed into LDS
// (same for B with N rows).  All four combinations are used by the layer
// catalog (fc fwd = NT-direct/direct, fc dx = direct/trans, dw = trans/trans).
//
// Structure: 128x128 tile, BK=32, 4 waves (2x2), 16x16x32 bf16 MFMA with
// fp32 accumulation; interior aligned tiles stage A/B via
// __builtin_amdgcn_global_load_lds (16 B/lane direct-to-LDS DMA); edge or
// unaligned tiles take a guarded staging path.  XCD-aware block swizzle for
// L2 locality; optional split-K with fp32 atomic reduction (STORE_MODE 2)
// for the skinny dw GEMMs.  Epilogue fuses per-column bias add + ReLU.
//
// Replaces the reference's cuBLAS calls inside conv/ip layers (SURVEY.md
// §3.6 rows "Convolution fwd/bwd", "InnerProduct") with a hand-written
// CDNA4 kernel per the rebuild's north star.

#include "common.h"

namespace cosamd {

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef unsigned short short8v __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

constexpr int BM = 128, BN = 128, BK = 32;
constexpr int NTHREADS = 256;

// LDS chunk swizzle: a row's four 16-byte k-chunks are stored rotated by
// XOR with ((row>>1)&3).  Without it, a fragment ds_read_b128's lane
// groups (16 lanes = 16 rows, same k-chunk) land on only 2 of the 8
// bank-quads (addr = row*64B + chunk*16B, 16*row mod 32 banks ∈ {0,16}),
// an 8-way conflict; the swizzle spreads them across all 8 quads (2-way,
// which CDNA4 serves conflict-free).  LDS stays linear so the
// global_load_lds fast path just pre-swizzles the per-lane SOURCE address
// (cdna_hip_programming.md §5: swizzle must be both-sides-or-neither).
__device__ __forceinline__ int swz_chunk(int row, int chunk) {
  return chunk ^ ((row >> 1) & 3);
}

// ---------------------------------------------------------------- staging

// direct ([rows][K]) guarded staging: vectorized 8-wide along K when the
// 16-byte slot is fully in-bounds, scalar otherwise; zero-fills the rest.
__device__ __forceinline__ void stage_direct_guarded(
    bf16* lds_, const bf16* g_, int row0, int rows, int ld,
    int k0, int kend, int tid, int nthreads = NTHREADS) {
  auto* lds = reinterpret_cast<unsigned short*>(lds_);
  auto* g = reinterpret_cast<const unsigned short*>(g_);
  // 4096 elements, 512 slots of 8
  for (int slot = tid; slot < 512; slot += nthreads) {
    int row = slot >> 2;            // 4 slots of 8 per 32-wide row
    int kk = (slot & 3) * 8;
    int grow = row0 + row;
    int gk = k0 + kk;
    unsigned short* dst = lds + row * BK + (swz_chunk(row, slot & 3) << 3);
    if (grow < rows && gk + 8 <= kend && ((ld | gk) % 8 == 0)) {
      *reinterpret_cast<short8v*>(dst) =
          *reinterpret_cast<const short8v*>(g + (int64_t)grow * ld + gk);
    } else if (grow < rows) {
#pragma unroll
      for (int j = 0; j < 8; ++j)
        dst[j] = (gk + j < kend) ? g[(int64_t)grow * ld + gk + j] : 0;
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) dst[j] = 0;
    }
  }
}

// ---- transposed ([K][rows]) operand staging into the CANONICAL layout
// with u32 k-pair packing: each slot loads the same 8 output-dim columns
// of TWO adjacent K-rows (two coalesced 16-byte global reads), packs
// (k, k+1) element pairs into u32s, and writes 8 ds_write_b32 — half
// the write instructions at 4x wider grain than the previous 16 scalar
// b16 scatters, and ~4-way instead of 16-way bank conflicts (row*64B
// strides alias banks; the k-pair dimension now spreads them).
// Fragment reads stay the standard swizzled b128 path.
__device__ __forceinline__ void stage_trans_pair_guarded(
    bf16* lds_, const bf16* g_, int row0, int rows, int ld,
    int k0, int kend, int tid, int nthreads = NTHREADS,
    int tile_rows = BM) {
  auto* lds = reinterpret_cast<unsigned short*>(lds_);
  auto* g = reinterpret_cast<const unsigned short*>(g_);
  int rslots = tile_rows >> 3;          // 8-column octets per k-pair
  int nslots = 16 * rslots;             // 16 k-pairs per BK=32 tile
  for (int slot = tid; slot < nslots; slot += nthreads) {
    int kp = slot & 15;                 // k-pair index (adjacent lanes
    int m0 = (slot >> 4) * 8;           //   get distinct kp: bank spread)
    unsigned short va[8], vb[8];
#pragma unroll
    for (int h = 0; h < 2; ++h) {
      unsigned short* v = h ? vb : va;
      int gk = k0 + kp * 2 + h;
      int gr = row0 + m0;
      if (gk < kend && gr + 8 <= rows && ((ld | gr) % 8 == 0)) {
        *reinterpret_cast<short8v*>(v) =
            *reinterpret_cast<const short8v*>(g + (int64_t)gk * ld + gr);
      } else if (gk < kend) {
#pragma unroll
        for (int j = 0; j < 8; ++j)
          v[j] = (gr + j < rows) ? g[(int64_t)gk * ld + gr + j] : 0;
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) v[j] = 0;
      }
    }
    int kk = kp * 2;
    int chunk0 = kk >> 3, kin = kk & 7;  // position inside a 16B chunk
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      int row = m0 + j;
      unsigned int packed = (unsigned int)va[j] |
                            ((unsigned int)vb[j] << 16);
      *reinterpret_cast<unsigned int*>(
          lds + row * BK + (swz_chunk(row, chunk0) << 3) + kin) = packed;
    }
  }
}

// --------------------------------------------------- wide-N dw kernel
// Specialized for the skinny trans/trans weight-gradient GEMMs
// (dw[Kout][Kcol] = dy^T @ col, K = N*P*Q huge): BN=256 halves the
// A-operand re-read factor vs the 128x128 kernel; fp32 atomic split-K.

constexpr int WBN = 256;
constexpr int WNT = 512;  // 8 waves: 2 (M) x 4 (N)

template <int STORE>  // 1 = plain fp32 store (no split-K), 2 = atomic
__global__ __launch_bounds__(WNT, 1) void gemm_tt_wide_kernel(
    const bf16* __restrict__ A, const bf16* __restrict__ B,
    float* __restrict__ C, int M, int N, int K, int lda, int ldb, int ldc,
    int ksplit, float alpha, float* __restrict__ db) {
  __shared__ bf16 As[BM * BK];
  __shared__ bf16 Bs[WBN * BK];

  int mblocks = (M + BM - 1) / BM;
  int nblocks = (N + WBN - 1) / WBN;
  int bid = xcd_swizzle(blockIdx.x, mblocks * nblocks);
  int bm = bid / nblocks, bn = bid % nblocks;
  int tile_m = bm * BM, tile_n = bn * WBN;
  int k_begin = blockIdx.y * ksplit;
  int k_end = min(K, k_begin + ksplit);

  int tid = threadIdx.x;
  int wave = tid >> 6, lane = tid & 63;
  int wm = wave >> 2, wn = wave & 3;     // 2x4 wave grid, 64x64 each
  int lrow = lane & 15, lk8 = (lane >> 4) * 8;

  f32x4 acc[4][4] = {};

  // fused bias gradient: db[m] += sum_k A[k][m] — the A tiles are in
  // LDS anyway; the bn==0 blocks of each k-split cover every k exactly
  // once (colsum was 4-5%% of the AlexNet/GoogLeNet steps as separate
  // ramp-bound launches over the same dy operand)
  bool do_db = (db != nullptr) && (bn == 0);
  for (int k0 = k_begin; k0 < k_end; k0 += BK) {
    stage_trans_pair_guarded(As, A, tile_m, M, lda, k0, k_end, tid, WNT,
                             BM);
    stage_trans_pair_guarded(Bs, B, tile_n, N, ldb, k0, k_end, tid, WNT,
                             WBN);
    __syncthreads();
    if (do_db && tid < BM && tile_m + tid < M) {
      float acc_b = 0.f;
      const bf16* row = As + tid * BK;
#pragma unroll
      for (int kk = 0; kk < BK; ++kk)
        acc_b += bf2f(row[(swz_chunk(tid, kk >> 3) << 3) | (kk & 7)]);
      atomicAdd(db + tile_m + tid, acc_b);
    }
    bf16x8 afrag[4], bfrag[4];
#pragma unroll
    for (int f = 0; f < 4; ++f) {
      int ra = wm * 64 + f * 16 + lrow;
      int rb = wn * 64 + f * 16 + lrow;
      afrag[f] = *reinterpret_cast<const bf16x8*>(
          As + ra * BK + (swz_chunk(ra, lk8 >> 3) << 3));
      bfrag[f] = *reinterpret_cast<const bf16x8*>(
          Bs + rb * BK + (swz_chunk(rb, lk8 >> 3) << 3));
    }
#pragma unroll
    for (int fm = 0; fm < 4; ++fm)
#pragma unroll
      for (int fn = 0; fn < 4; ++fn)
        acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afrag[fm], bfrag[fn], acc[fm][fn], 0, 0, 0);
    __syncthreads();
  }

  int crow0 = tile_m + wm * 64 + ((lane >> 4) << 2);
  int ccol0 = tile_n + wn * 64 + (lane & 15);
#pragma unroll
  for (int fm = 0; fm < 4; ++fm)
#pragma unroll
    for (int fn = 0; fn < 4; ++fn) {
      int col = ccol0 + fn * 16;
      if (col >= N) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = crow0 + fm * 16 + r;
        if (row >= M) continue;
        if (STORE == 1)
          C[(int64_t)row * ldc + col] = acc[fm][fn][r] * alpha;
        else
          atomicAdd(C + (int64_t)row * ldc + col, acc[fm][fn][r] * alpha);
      }
    }
}

// fast path: interior tile, aligned — direct-to-LDS DMA, 16 B per lane.
template <int WAVES = 4>
__device__ __forceinline__ void stage_direct_fast(
    bf16* lds, const bf16* g, int row0, int ld, int k0, int tid) {
  int wave = tid >> 6, lane = tid & 63;
  constexpr int CPW = 8 / WAVES;         // chunks per wave
#pragma unroll
  for (int it = 0; it < CPW; ++it) {
    int chunk = wave * CPW + it;         // 8 chunks of 512 elements
    int idx = chunk * 512 + lane * 8;    // linear element index in tile
    int row = idx >> 5;                  // /32
    int kk = swz_chunk(row, (idx & 31) >> 3) << 3;  // pre-swizzled source
    auto* gp = (const __attribute__((address_space(1))) unsigned int*)(
        g + (int64_t)(row0 + row) * ld + k0 + kk);
    auto* lp = (__attribute__((address_space(3))) unsigned int*)(
        lds + chunk * 512);
    __builtin_amdgcn_global_load_lds(gp, lp, 16, 0, 0);
  }
}

// ------------------------------------------------------------------ kernel

// WAVES=4: 2x2 wave grid of 64x64 (2 blocks/CU = 8 waves/CU);
// WAVES=8: 4x2 grid of 32x64 (2 blocks/CU = 16 waves/CU — doubles the
// wave-level parallelism hiding the staging/fragment latency).
template <bool TRANS_A, bool TRANS_B, int STORE_MODE, int WAVES = 4>
__global__ __launch_bounds__(WAVES * 64, 2) void gemm_kernel(
    const bf16* __restrict__ A, const bf16* __restrict__ B,
    void* __restrict__ C, const float* __restrict__ bias,
    int M, int N, int K, int lda, int ldb, int ldc,
    int ksplit, int relu, float alpha, int m_alloc, int n_alloc,
    int64_t sA, int64_t sB, int64_t sC) {
  // batched operation (blockIdx.z = batch index, strides in elements for
  // A/B and BYTES for C since its type depends on STORE_MODE)
  A += (int64_t)blockIdx.z * sA;
  B += (int64_t)blockIdx.z * sB;
  C = (void*)((char*)C + (int64_t)blockIdx.z * sC);
  __shared__ bf16 Asb[2][BM * BK];
  __shared__ bf16 Bsb[2][BN * BK];

  int mblocks = (M + BM - 1) / BM;
  int nblocks = (N + BN - 1) / BN;
  int bid = xcd_swizzle(blockIdx.x, mblocks * nblocks);
  int bm = bid / nblocks, bn = bid % nblocks;
  int tile_m = bm * BM, tile_n = bn * BN;

  int k_begin = blockIdx.y * ksplit;
  int k_end = min(K, k_begin + ksplit);

  constexpr int NT = WAVES * 64;
  constexpr int MROWS = BM / (WAVES / 2);   // rows per wave (64 or 32)
  constexpr int MF = MROWS / 16;            // m-fragments per wave
  int tid = threadIdx.x;
  int wave = tid >> 6, lane = tid & 63;
  int wm = wave >> 1, wn = wave & 1;     // (WAVES/2) x 2 wave grid
  int lrow = lane & 15, lk8 = (lane >> 4) * 8;

  f32x4 acc[MF][4] = {};

  // fast path only needs the STAGED rows to be readable: callers may
  // over-allocate operand buffers (m_alloc/n_alloc >= M/N) so edge tiles
  // stage junk rows whose outputs the epilogue guards discard
  bool a_fast = !TRANS_A && (tile_m + BM <= m_alloc) && (lda % 8 == 0) &&
                ((k_end - k_begin) % BK == 0) && (k_begin % 8 == 0);
  bool b_fast = !TRANS_B && (tile_n + BN <= n_alloc) && (ldb % 8 == 0) &&
                ((k_end - k_begin) % BK == 0) && (k_begin % 8 == 0);

  if (a_fast && b_fast && (k_end - k_begin) >= 2 * BK) {
    // Pipelined double-buffered path (interior aligned tiles): the next
    // K-tile's direct-to-LDS DMA is issued BEFORE computing the current
    // tile, and the pre-barrier wait is a COUNTED `s_waitcnt vmcnt(4)`
    // (the 4 loads just issued stay in flight across the barrier) rather
    // than __syncthreads' full drain — cdna_hip_programming.md §5.5
    // T3/T4: "never drain vmcnt to 0 in the main loop".
    int cur = 0;
    stage_direct_fast<WAVES>(Asb[0], A, tile_m, lda, k_begin, tid);
    stage_direct_fast<WAVES>(Bsb[0], B, tile_n, ldb, k_begin, tid);
    for (int k0 = k_begin; k0 < k_end; k0 += BK) {
      if (k0 + BK < k_end) {
        stage_direct_fast<WAVES>(Asb[cur ^ 1], A, tile_m, lda, k0 + BK,
                                 tid);
        stage_direct_fast<WAVES>(Bsb[cur ^ 1], B, tile_n, ldb, k0 + BK,
                                 tid);
      }
      // wait for the CURRENT tile's DMA loads; the prefetched tile's
      // (2 per operand per wave at WAVES=4, 1 each at 8) remain
      // outstanding
      if (WAVES == 4)
        asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
      else
        asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
      __builtin_amdgcn_sched_barrier(0);
      __builtin_amdgcn_s_barrier();
      const bf16* Asp = Asb[cur];
      const bf16* Bsp = Bsb[cur];
      bf16x8 afrag[MF], bfrag[4];
#pragma unroll
      for (int f = 0; f < 4; ++f) {
        int rb = wn * 64 + f * 16 + lrow;
        bfrag[f] = *reinterpret_cast<const bf16x8*>(
            Bsp + rb * BK + (swz_chunk(rb, lk8 >> 3) << 3));
      }
#pragma unroll
      for (int f = 0; f < MF; ++f) {
        int ra = wm * MROWS + f * 16 + lrow;
        afrag[f] = *reinterpret_cast<const bf16x8*>(
            Asp + ra * BK + (swz_chunk(ra, lk8 >> 3) << 3));
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int fm = 0; fm < MF; ++fm)
#pragma unroll
        for (int fn = 0; fn < 4; ++fn)
          acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[fm], bfrag[fn], acc[fm][fn], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
      // all waves finish reading buf[cur] before the next iteration's
      // stage overwrites it; the ds_reads completed before their MFMAs,
      // so no counter drain is needed at this barrier
      __builtin_amdgcn_s_barrier();
      cur ^= 1;
    }
  } else {
    bf16* As = Asb[0];
    bf16* Bs = Bsb[0];
    for (int k0 = k_begin; k0 < k_end; k0 += BK) {
      if (a_fast) {
        stage_direct_fast<WAVES>(As, A, tile_m, lda, k0, tid);
      } else if (TRANS_A) {
        stage_trans_pair_guarded(As, A, tile_m, M, lda, k0, k_end, tid,
                                 NT);
      } else {
        stage_direct_guarded(As, A, tile_m, M, lda, k0, k_end, tid, NT);
      }
      if (b_fast) {
        stage_direct_fast<WAVES>(Bs, B, tile_n, ldb, k0, tid);
      } else if (TRANS_B) {
        stage_trans_pair_guarded(Bs, B, tile_n, N, ldb, k0, k_end, tid,
                                 NT);
      } else {
        stage_direct_guarded(Bs, B, tile_n, N, ldb, k0, k_end, tid, NT);
      }
      __syncthreads();

      bf16x8 afrag[MF], bfrag[4];
#pragma unroll
      for (int f = 0; f < 4; ++f) {
        int rb = wn * 64 + f * 16 + lrow;
        bfrag[f] = *reinterpret_cast<const bf16x8*>(
            Bs + rb * BK + (swz_chunk(rb, lk8 >> 3) << 3));
      }
#pragma unroll
      for (int f = 0; f < MF; ++f) {
        int ra = wm * MROWS + f * 16 + lrow;
        afrag[f] = *reinterpret_cast<const bf16x8*>(
            As + ra * BK + (swz_chunk(ra, lk8 >> 3) << 3));
      }
#pragma unroll
      for (int fm = 0; fm < MF; ++fm)
#pragma unroll
        for (int fn = 0; fn < 4; ++fn)
          acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[fm], bfrag[fn], acc[fm][fn], 0, 0, 0);

      __syncthreads();
    }
  }

  // ------------------------------------------------------------- epilogue
  int crow0 = tile_m + wm * MROWS + ((lane >> 4) << 2);
  int ccol0 = tile_n + wn * 64 + (lane & 15);
#pragma unroll
  for (int fm = 0; fm < MF; ++fm) {
#pragma unroll
    for (int fn = 0; fn < 4; ++fn) {
      int col = ccol0 + fn * 16;
      if (col >= N) continue;
      float badd = (bias != nullptr) ? bias[col] : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = crow0 + fm * 16 + r;
        if (row >= M) continue;
        float v = acc[fm][fn][r] * alpha + badd;
        if (relu && v < 0.f) v = 0.f;
        int64_t off = (int64_t)row * ldc + col;
        if (STORE_MODE == 0) {
          reinterpret_cast<bf16*>(C)[off] = f2bf(v);
        } else if (STORE_MODE == 3) {
          // bf16 read-modify-write accumulate (dx fan-in; exclusive
          // tiles, single split -> race-free)
          bf16* p = reinterpret_cast<bf16*>(C) + off;
          *p = f2bf(v + bf2f(*p));
        } else if (STORE_MODE == 1) {
          reinterpret_cast<float*>(C)[off] = v;
        } else {
          atomicAdd(reinterpret_cast<float*>(C) + off, v);
        }
      }
    }
  }
}

// ------------------------------------------ implicit-im2col conv GEMMs
// The col matrix's 8-element k-runs (one (r,s) tap, 8 consecutive
// channels) are contiguous channel spans of the NHWC input whenever
// Cg % 8 == 0, so the GEMM stages its col operand straight from x —
// zero-filling padding taps — and the col matrix (AlexNet conv2 alone:
// ~900 MB written + read twice per step) never materializes.

struct CGeom {
  int H, W, C;                       // input NHWC dims (C = all channels)
  int Q, PQ;                         // output width, P*Q
  int sh, sw, ph, pw, dil, S;
  int c0, Cg;                        // group channel window
  int Kcol;                          // R*S*Cg (zero beyond)
  // magic reciprocals: floor(n/d) == (n * m) >> 40 for n < 2^20
  unsigned long long mPQ, mQ, mCg, mS;
};

__device__ __forceinline__ unsigned mdiv(unsigned n,
                                         unsigned long long m) {
  return (unsigned)(((unsigned long long)n * m) >> 40);
}

static inline unsigned long long magic40(unsigned d) {
  return ((1ull << 40) + d - 1) / d;
}

// per-block output-pixel decode: rinfo[row] = {n*H*W, p*sh-ph, q*sw-pw}
__device__ __forceinline__ void decode_rows(int* rinfo, const CGeom& gm,
                                            int tile_m, int M, int rows,
                                            int tid) {
  for (int i = tid; i < rows; i += blockDim.x) {
    int m = tile_m + i;
    if (m >= M) m = M - 1;            // junk rows: valid addrs, masked out
    unsigned n = mdiv(m, gm.mPQ);
    unsigned pq = m - n * gm.PQ;
    unsigned p = mdiv(pq, gm.mQ);
    unsigned q = pq - p * gm.Q;
    rinfo[i * 3] = (int)(n * gm.H * gm.W);
    rinfo[i * 3 + 1] = (int)p * gm.sh - gm.ph;
    rinfo[i * 3 + 2] = (int)q * gm.sw - gm.pw;
  }
}

// A-operand staging for the forward conv GEMM: same LDS layout and DMA
// placement as stage_direct_fast, but the per-lane source address is the
// im2col gather.  Out-of-image (or k >= Kcol) lanes redirect their DMA
// to a 16-byte zero page instead of branching: the instruction always
// issues, so the pipeline's counted s_waitcnt vmcnt(N) stays exact (a
// divergent skip would desynchronize the count), and no LDS zero-fill
// writes are needed.
template <int WAVES>
__device__ __forceinline__ void stage_implicit_fast(
    bf16* lds_, const bf16* X, const bf16* zpage, const CGeom& gm,
    const int* rinfo, int k0, int tid) {
  auto* lds = reinterpret_cast<unsigned short*>(lds_);
  int wave = tid >> 6, lane = tid & 63;
  constexpr int CPW = 8 / WAVES;
#pragma unroll
  for (int it = 0; it < CPW; ++it) {
    int chunk = wave * CPW + it;
    int idx = chunk * 512 + lane * 8;
    int row = idx >> 5;
    int kk = swz_chunk(row, (idx & 31) >> 3) << 3;
    int k = k0 + kk;
    unsigned rs = mdiv(k, gm.mCg);
    int c = k - rs * gm.Cg;
    unsigned r = mdiv(rs, gm.mS);
    int s = rs - r * gm.S;
    int h = rinfo[row * 3 + 1] + (int)r * gm.dil;
    int w = rinfo[row * 3 + 2] + s * gm.dil;
    bool ok = (k < gm.Kcol) && (h >= 0) && (h < gm.H) && (w >= 0) &&
              (w < gm.W);
    int64_t off = ((int64_t)rinfo[row * 3] +
                   (int64_t)(h * gm.W + w)) * gm.C + gm.c0 + c;
    const bf16* src = ok ? X + off : zpage;
    auto* lp = (__attribute__((address_space(3))) unsigned int*)(
        lds + chunk * 512);
    auto* gp = (const __attribute__((address_space(1))) unsigned int*)src;
    __builtin_amdgcn_global_load_lds(gp, lp, 16, 0, 0);
  }
}

// B-operand staging for the dw trans/trans GEMM: canonical u32 k-pair
// layout like stage_trans_pair_guarded, but each 8-column octet (one
// (r,s) tap, 8 channels) is gathered from x for two adjacent output
// pixels (the k-pair).
__device__ __forceinline__ void stage_trans_pair_implicit(
    bf16* lds_, const bf16* X, const CGeom& gm, int row0,
    int k0, int kend, int tid, int nthreads, int tile_rows) {
  auto* lds = reinterpret_cast<unsigned short*>(lds_);
  int rslots = tile_rows >> 3;
  int nslots = 16 * rslots;
  for (int slot = tid; slot < nslots; slot += nthreads) {
    int kp = slot & 15;
    int m0 = (slot >> 4) * 8;
    int gr = row0 + m0;                 // col-matrix column octet base
    unsigned rs = mdiv(gr, gm.mCg);
    int c = gr - rs * gm.Cg;
    unsigned r = mdiv(rs, gm.mS);
    int s = rs - r * gm.S;
    bool col_ok = gr < gm.Kcol;         // octet fully in/out (Kcol%8==0)
    int hr = (int)r * gm.dil - gm.ph;
    int ws = s * gm.dil - gm.pw;
    unsigned short va[8] = {}, vb[8] = {};
#pragma unroll
    for (int hh = 0; hh < 2; ++hh) {
      unsigned short* v = hh ? vb : va;
      int gk = k0 + kp * 2 + hh;        // output-pixel (npq) index
      if (gk < kend && col_ok) {
        unsigned n = mdiv(gk, gm.mPQ);
        unsigned pq = gk - n * gm.PQ;
        unsigned p = mdiv(pq, gm.mQ);
        unsigned q = pq - p * gm.Q;
        int hi = (int)p * gm.sh + hr;
        int wi = (int)q * gm.sw + ws;
        if (hi >= 0 && hi < gm.H && wi >= 0 && wi < gm.W) {
          const unsigned short* src =
              reinterpret_cast<const unsigned short*>(X) +
              ((int64_t)(n * gm.H + hi) * gm.W + wi) * gm.C + gm.c0 + c;
          *reinterpret_cast<short8v*>(v) =
              *reinterpret_cast<const short8v*>(src);
        }
      }
    }
    int kk = kp * 2;
    int chunk0 = kk >> 3, kin = kk & 7;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      int row = m0 + j;
      unsigned int packed = (unsigned int)va[j] |
                            ((unsigned int)vb[j] << 16);
      *reinterpret_cast<unsigned int*>(
          lds + row * BK + (swz_chunk(row, chunk0) << 3) + kin) = packed;
    }
  }
}

// ---- small-C implicit staging (conv1-class layers, C % 8 != 0).
// With dil==1, c0==0 and full channels, the col matrix's k axis within
// one filter row r is a CONTIGUOUS span of the input row (k = r*S*C +
// s*C + c), so an 8-element slot is one unaligned 16-byte load (gfx950
// supports it natively), merging two loads when the slot straddles a
// filter-row boundary (S*C >= 8 keeps it to at most one straddle).
// CGeom field reuse for the _sc kernels: Cg/mCg hold S*C, S/mS hold C.

// NN A-operand [row=npq][k] into the canonical swizzled layout
__device__ __forceinline__ void stage_implicit_sc(
    bf16* lds_, const bf16* X, const CGeom& gm, const int* rinfo,
    int k0, int tid, int nthreads) {
  auto* lds = reinterpret_cast<unsigned short*>(lds_);
  auto* x = reinterpret_cast<const unsigned short*>(X);
  int SC = gm.Cg, C = gm.S;
  for (int slot = tid; slot < 512; slot += nthreads) {
    int row = slot >> 2;
    int kk = (slot & 3) * 8;
    int k = k0 + kk;
    unsigned short* dst = lds + row * BK + (swz_chunk(row, slot & 3) << 3);
    unsigned r0 = mdiv(k, gm.mCg);          // filter row
    int t0 = k - r0 * SC;                   // s*C + c within the row
    int h = rinfo[row * 3 + 1] + (int)r0;
    int w0 = rinfo[row * 3 + 2];
    int sp = SC - t0;                       // elements before r0+1
    // fully-interior slot: the spanned w range [w0, w0+S) is in-image
    bool interior = k + 8 <= gm.Kcol && h >= 0 && w0 >= 0 &&
                    w0 * C + SC <= gm.W * C &&
                    (sp >= 8 ? h < gm.H : h + 1 < gm.H);
    if (interior) {
      const unsigned short* a0 =
          x + ((int64_t)rinfo[row * 3] + (int64_t)h * gm.W) * C +
          w0 * C + t0;
      if (sp >= 8) {
        __builtin_memcpy(dst, a0, 16);
      } else {
        unsigned short lo[8], hi[8];
        __builtin_memcpy(lo, a0, 16);
        const unsigned short* a1 =
            x + ((int64_t)rinfo[row * 3] + (int64_t)(h + 1) * gm.W) * C +
            w0 * C;
        __builtin_memcpy(hi, a1, 16);
#pragma unroll
        for (int j = 0; j < 8; ++j) dst[j] = j < sp ? lo[j] : hi[j - sp];
      }
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        int kj = k + j;
        unsigned short v = 0;
        if (kj < gm.Kcol) {
          unsigned r = mdiv(kj, gm.mCg);
          int t = kj - r * SC;
          unsigned s = mdiv(t, gm.mS);
          int c = t - s * C;
          int hh = rinfo[row * 3 + 1] + (int)r;
          int ww = w0 + (int)s;
          if (hh >= 0 && hh < gm.H && ww >= 0 && ww < gm.W)
            v = x[((int64_t)rinfo[row * 3] +
                   (int64_t)hh * gm.W + ww) * C + c];
        }
        dst[j] = v;
      }
    }
  }
}

// TT B-operand [k=npq][col] u32 k-pair canonical layout
__device__ __forceinline__ void stage_trans_pair_implicit_sc(
    bf16* lds_, const bf16* X, const CGeom& gm, int row0,
    int k0, int kend, int tid, int nthreads, int tile_rows) {
  auto* lds = reinterpret_cast<unsigned short*>(lds_);
  auto* x = reinterpret_cast<const unsigned short*>(X);
  int SC = gm.Cg, C = gm.S;
  int rslots = tile_rows >> 3;
  int nslots = 16 * rslots;
  for (int slot = tid; slot < nslots; slot += nthreads) {
    int kp = slot & 15;
    int m0 = (slot >> 4) * 8;
    int gr = row0 + m0;                  // col-matrix column octet base
    unsigned r0 = mdiv(gr, gm.mCg);
    int t0 = gr - r0 * SC;
    int sp = SC - t0;
    unsigned short va[8] = {}, vb[8] = {};
#pragma unroll
    for (int hh2 = 0; hh2 < 2; ++hh2) {
      unsigned short* v = hh2 ? vb : va;
      int gk = k0 + kp * 2 + hh2;        // output-pixel index
      if (gk >= kend) continue;
      unsigned n = mdiv(gk, gm.mPQ);
      unsigned pq = gk - n * gm.PQ;
      unsigned p = mdiv(pq, gm.mQ);
      unsigned q = pq - p * gm.Q;
      int h = (int)p * gm.sh - gm.ph + (int)r0;
      int w0 = (int)q * gm.sw - gm.pw;
      bool interior = gr + 8 <= gm.Kcol && h >= 0 && w0 >= 0 &&
                      w0 * C + SC <= gm.W * C &&
                      (sp >= 8 ? h < gm.H : h + 1 < gm.H);
      int64_t nHW = (int64_t)(n * gm.H);
      if (interior) {
        const unsigned short* a0 =
            x + (nHW + h) * (int64_t)gm.W * C + w0 * C + t0;
        if (sp >= 8) {
          __builtin_memcpy(v, a0, 16);
        } else {
          unsigned short lo[8], hi[8];
          __builtin_memcpy(lo, a0, 16);
          const unsigned short* a1 =
              x + (nHW + h + 1) * (int64_t)gm.W * C + w0 * C;
          __builtin_memcpy(hi, a1, 16);
#pragma unroll
          for (int j = 0; j < 8; ++j) v[j] = j < sp ? lo[j] : hi[j - sp];
        }
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          int kj = gr + j;
          if (kj >= gm.Kcol) continue;
          unsigned r = mdiv(kj, gm.mCg);
          int t = kj - r * SC;
          unsigned s = mdiv(t, gm.mS);
          int c = t - s * C;
          int hh = (int)p * gm.sh - gm.ph + (int)r;
          int ww = w0 + (int)s;
          if (hh >= 0 && hh < gm.H && ww >= 0 && ww < gm.W)
            v[j] = x[(nHW + hh) * (int64_t)gm.W * C + ww * C + c];
        }
      }
    }
    int kk = kp * 2;
    int chunk0 = kk >> 3, kin = kk & 7;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      int row = m0 + j;
      unsigned int packed = (unsigned int)va[j] |
                            ((unsigned int)vb[j] << 16);
      *reinterpret_cast<unsigned int*>(
          lds + row * BK + (swz_chunk(row, chunk0) << 3) + kin) = packed;
    }
  }
}

// small-C forward conv GEMM: double-buffered __syncthreads pipeline
// (the ds_write staging has no counted-vmcnt scheme to preserve)
template <int WAVES>
__global__ __launch_bounds__(WAVES * 64, 2) void gemm_conv_fwd_sc_kernel(
    const bf16* __restrict__ X, const bf16* __restrict__ B,
    bf16* __restrict__ C, const float* __restrict__ bias,
    int M, int N, int K, int ldb, int ldc, int relu, CGeom gm) {
  __shared__ bf16 Asb[2][BM * BK];
  __shared__ bf16 Bsb[2][BN * BK];
  __shared__ int rinfo[BM * 3];

  int mblocks = (M + BM - 1) / BM;
  int nblocks = (N + BN - 1) / BN;
  int bid = xcd_swizzle(blockIdx.x, mblocks * nblocks);
  int bm = bid / nblocks, bn = bid % nblocks;
  int tile_m = bm * BM, tile_n = bn * BN;

  constexpr int NT = WAVES * 64;
  constexpr int MROWS = BM / (WAVES / 2);
  constexpr int MF = MROWS / 16;
  int tid = threadIdx.x;
  int wave = tid >> 6, lane = tid & 63;
  int wm = wave >> 1, wn = wave & 1;
  int lrow = lane & 15, lk8 = (lane >> 4) * 8;

  decode_rows(rinfo, gm, tile_m, M, BM, tid);
  __syncthreads();

  f32x4 acc[MF][4] = {};
  // double-buffered, ONE barrier per tile: staging t+1 (into buf^1)
  // overlaps compute of t; the end-of-iteration barrier covers both
  // "staging t+1 done" and "all reads of buf[t&1] done" before the
  // next iteration's stage(t+2) overwrites it
  int T = K / BK;
  stage_implicit_sc(Asb[0], X, gm, rinfo, 0, tid, NT);
  stage_direct_fast<WAVES>(Bsb[0], B, tile_n, ldb, 0, tid);
  __syncthreads();
  for (int t = 0; t < T; ++t) {
    if (t + 1 < T) {
      stage_implicit_sc(Asb[(t + 1) & 1], X, gm, rinfo, (t + 1) * BK,
                        tid, NT);
      stage_direct_fast<WAVES>(Bsb[(t + 1) & 1], B, tile_n, ldb,
                               (t + 1) * BK, tid);
    }
    const bf16* Asp = Asb[t & 1];
    const bf16* Bsp = Bsb[t & 1];
    bf16x8 afrag[MF], bfrag[4];
#pragma unroll
    for (int f = 0; f < 4; ++f) {
      int rb = wn * 64 + f * 16 + lrow;
      bfrag[f] = *reinterpret_cast<const bf16x8*>(
          Bsp + rb * BK + (swz_chunk(rb, lk8 >> 3) << 3));
    }
#pragma unroll
    for (int f = 0; f < MF; ++f) {
      int ra = wm * MROWS + f * 16 + lrow;
      afrag[f] = *reinterpret_cast<const bf16x8*>(
          Asp + ra * BK + (swz_chunk(ra, lk8 >> 3) << 3));
    }
#pragma unroll
    for (int fm = 0; fm < MF; ++fm)
#pragma unroll
      for (int fn = 0; fn < 4; ++fn)
        acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afrag[fm], bfrag[fn], acc[fm][fn], 0, 0, 0);
    __syncthreads();
  }

  int crow0 = tile_m + wm * MROWS + ((lane >> 4) << 2);
  int ccol0 = tile_n + wn * 64 + (lane & 15);
#pragma unroll
  for (int fm = 0; fm < MF; ++fm) {
#pragma unroll
    for (int fn = 0; fn < 4; ++fn) {
      int col = ccol0 + fn * 16;
      if (col >= N) continue;
      float badd = (bias != nullptr) ? bias[col] : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = crow0 + fm * 16 + r;
        if (row >= M) continue;
        float v = acc[fm][fn][r] + badd;
        if (relu && v < 0.f) v = 0.f;
        C[(int64_t)row * ldc + col] = f2bf(v);
      }
    }
  }
}

// small-C dw trans/trans GEMM (B operand gathered contiguously from x)
template <int STORE>
__global__ __launch_bounds__(WNT, 1) void gemm_conv_dw_sc_kernel(
    const bf16* __restrict__ A, const bf16* __restrict__ X,
    float* __restrict__ C, int M, int N, int K, int lda, int ldc,
    int ksplit, float alpha, float* __restrict__ db, CGeom gm) {
  __shared__ bf16 As[BM * BK];
  __shared__ bf16 Bs[WBN * BK];

  int mblocks = (M + BM - 1) / BM;
  int nblocks = (N + WBN - 1) / WBN;
  int bid = xcd_swizzle(blockIdx.x, mblocks * nblocks);
  int bm = bid / nblocks, bn = bid % nblocks;
  int tile_m = bm * BM, tile_n = bn * WBN;
  int k_begin = blockIdx.y * ksplit;
  int k_end = min(K, k_begin + ksplit);

  int tid = threadIdx.x;
  int wave = tid >> 6, lane = tid & 63;
  int wm = wave >> 2, wn = wave & 3;
  int lrow = lane & 15, lk8 = (lane >> 4) * 8;

  f32x4 acc[4][4] = {};
  bool do_db = (db != nullptr) && (bn == 0);
  for (int k0 = k_begin; k0 < k_end; k0 += BK) {
    stage_trans_pair_guarded(As, A, tile_m, M, lda, k0, k_end, tid, WNT,
                             BM);
    stage_trans_pair_implicit_sc(Bs, X, gm, tile_n, k0, k_end, tid, WNT,
                                 WBN);
    __syncthreads();
    if (do_db && tid < BM && tile_m + tid < M) {
      float acc_b = 0.f;
      const bf16* row = As + tid * BK;
#pragma unroll
      for (int kk = 0; kk < BK; ++kk)
        acc_b += bf2f(row[(swz_chunk(tid, kk >> 3) << 3) | (kk & 7)]);
      atomicAdd(db + tile_m + tid, acc_b);
    }
    bf16x8 afrag[4], bfrag[4];
#pragma unroll
    for (int f = 0; f < 4; ++f) {
      int ra = wm * 64 + f * 16 + lrow;
      int rb = wn * 64 + f * 16 + lrow;
      afrag[f] = *reinterpret_cast<const bf16x8*>(
          As + ra * BK + (swz_chunk(ra, lk8 >> 3) << 3));
      bfrag[f] = *reinterpret_cast<const bf16x8*>(
          Bs + rb * BK + (swz_chunk(rb, lk8 >> 3) << 3));
    }
#pragma unroll
    for (int fm = 0; fm < 4; ++fm)
#pragma unroll
      for (int fn = 0; fn < 4; ++fn)
        acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afrag[fm], bfrag[fn], acc[fm][fn], 0, 0, 0);
    __syncthreads();
  }

  int crow0 = tile_m + wm * 64 + ((lane >> 4) << 2);
  int ccol0 = tile_n + wn * 64 + (lane & 15);
#pragma unroll
  for (int fm = 0; fm < 4; ++fm)
#pragma unroll
    for (int fn = 0; fn < 4; ++fn) {
      int col = ccol0 + fn * 16;
      if (col >= N) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = crow0 + fm * 16 + r;
        if (row >= M) continue;
        if (STORE == 1)
          C[(int64_t)row * ldc + col] = acc[fm][fn][r] * alpha;
        else
          atomicAdd(C + (int64_t)row * ldc + col, acc[fm][fn][r] * alpha);
      }
    }
}

// forward conv GEMM: C[NPQ, Kg] = im2col(x) @ Wr^T, bias+ReLU fused.
// Always the pipelined double-buffered loop: the implicit A staging
// zero-fills k >= Kcol, so K runs over Kpad (32-aligned by the host)
// with no guarded edge cases, and Wr's padded rows/columns are zero.
template <int WAVES>
__global__ __launch_bounds__(WAVES * 64, 2) void gemm_conv_fwd_kernel(
    const bf16* __restrict__ X, const bf16* __restrict__ B,
    bf16* __restrict__ C, const float* __restrict__ bias,
    const bf16* __restrict__ zpage,
    int M, int N, int K, int ldb, int ldc, int relu, int accum,
    CGeom gm) {
  __shared__ bf16 Asb[2][BM * BK];
  __shared__ bf16 Bsb[2][BN * BK];
  __shared__ int rinfo[BM * 3];

  int mblocks = (M + BM - 1) / BM;
  int nblocks = (N + BN - 1) / BN;
  int bid = xcd_swizzle(blockIdx.x, mblocks * nblocks);
  int bm = bid / nblocks, bn = bid % nblocks;
  int tile_m = bm * BM, tile_n = bn * BN;

  constexpr int NT = WAVES * 64;
  constexpr int MROWS = BM / (WAVES / 2);
  constexpr int MF = MROWS / 16;
  int tid = threadIdx.x;
  int wave = tid >> 6, lane = tid & 63;
  int wm = wave >> 1, wn = wave & 1;
  int lrow = lane & 15, lk8 = (lane >> 4) * 8;

  decode_rows(rinfo, gm, tile_m, M, BM, tid);
  __syncthreads();

  f32x4 acc[MF][4] = {};
  int cur = 0;
  stage_implicit_fast<WAVES>(Asb[0], X, zpage, gm, rinfo, 0, tid);
  stage_direct_fast<WAVES>(Bsb[0], B, tile_n, ldb, 0, tid);
  for (int k0 = 0; k0 < K; k0 += BK) {
    if (k0 + BK < K) {
      stage_implicit_fast<WAVES>(Asb[cur ^ 1], X, zpage, gm, rinfo,
                                 k0 + BK, tid);
      stage_direct_fast<WAVES>(Bsb[cur ^ 1], B, tile_n, ldb, k0 + BK,
                               tid);
    }
    if (WAVES == 4)
      asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    else
      asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
    __builtin_amdgcn_sched_barrier(0);
    __builtin_amdgcn_s_barrier();
    const bf16* Asp = Asb[cur];
    const bf16* Bsp = Bsb[cur];
    bf16x8 afrag[MF], bfrag[4];
#pragma unroll
    for (int f = 0; f < 4; ++f) {
      int rb = wn * 64 + f * 16 + lrow;
      bfrag[f] = *reinterpret_cast<const bf16x8*>(
          Bsp + rb * BK + (swz_chunk(rb, lk8 >> 3) << 3));
    }
#pragma unroll
    for (int f = 0; f < MF; ++f) {
      int ra = wm * MROWS + f * 16 + lrow;
      afrag[f] = *reinterpret_cast<const bf16x8*>(
          Asp + ra * BK + (swz_chunk(ra, lk8 >> 3) << 3));
    }
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int fm = 0; fm < MF; ++fm)
#pragma unroll
      for (int fn = 0; fn < 4; ++fn)
        acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afrag[fm], bfrag[fn], acc[fm][fn], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();
    cur ^= 1;
  }

  int crow0 = tile_m + wm * MROWS + ((lane >> 4) << 2);
  int ccol0 = tile_n + wn * 64 + (lane & 15);
#pragma unroll
  for (int fm = 0; fm < MF; ++fm) {
#pragma unroll
    for (int fn = 0; fn < 4; ++fn) {
      int col = ccol0 + fn * 16;
      if (col >= N) continue;
      float badd = (bias != nullptr) ? bias[col] : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = crow0 + fm * 16 + r;
        if (row >= M) continue;
        float v = acc[fm][fn][r] + badd;
        if (relu && v < 0.f) v = 0.f;
        int64_t off = (int64_t)row * ldc + col;
        if (accum) v += bf2f(C[off]);   // dx fan-in: exclusive tiles,
        C[off] = f2bf(v);               // so plain RMW is race-free
      }
    }
  }
}

// quad-buffered variant: four LDS buffers (65.5 KB, still 2 blocks/CU
// within CDNA4's 160 KB) let ONE barrier per k-tile replace the usual
// two — the buffer written at iter t+1 (slot (t+3)&3) was last read at
// iter t-1, and barrier_t separates them (a wave's ds_reads complete
// before its MFMAs consume them, hence before it arrives at barrier_t).
template <int WAVES>
__global__ __launch_bounds__(WAVES * 64, 2) void gemm_conv_fwd_q_kernel(
    const bf16* __restrict__ X, const bf16* __restrict__ B,
    bf16* __restrict__ C, const float* __restrict__ bias,
    const bf16* __restrict__ zpage,
    int M, int N, int K, int ldb, int ldc, int relu, int accum,
    CGeom gm) {
  __shared__ bf16 Asb[4][BM * BK];
  __shared__ bf16 Bsb[4][BN * BK];
  __shared__ int rinfo[BM * 3];

  int mblocks = (M + BM - 1) / BM;
  int nblocks = (N + BN - 1) / BN;
  int bid = xcd_swizzle(blockIdx.x, mblocks * nblocks);
  int bm = bid / nblocks, bn = bid % nblocks;
  int tile_m = bm * BM, tile_n = bn * BN;

  constexpr int MROWS = BM / (WAVES / 2);
  constexpr int MF = MROWS / 16;
  int tid = threadIdx.x;
  int wave = tid >> 6, lane = tid & 63;
  int wm = wave >> 1, wn = wave & 1;
  int lrow = lane & 15, lk8 = (lane >> 4) * 8;

  decode_rows(rinfo, gm, tile_m, M, BM, tid);
  __syncthreads();

  f32x4 acc[MF][4] = {};
  int T = K / BK;
  stage_implicit_fast<WAVES>(Asb[0], X, zpage, gm, rinfo, 0, tid);
  stage_direct_fast<WAVES>(Bsb[0], B, tile_n, ldb, 0, tid);
  if (T > 1) {
    stage_implicit_fast<WAVES>(Asb[1], X, zpage, gm, rinfo, BK, tid);
    stage_direct_fast<WAVES>(Bsb[1], B, tile_n, ldb, BK, tid);
  }
  for (int t = 0; t < T; ++t) {
    if (t + 2 < T) {
      stage_implicit_fast<WAVES>(Asb[(t + 2) & 3], X, zpage, gm, rinfo,
                                 (t + 2) * BK, tid);
      stage_direct_fast<WAVES>(Bsb[(t + 2) & 3], B, tile_n, ldb,
                               (t + 2) * BK, tid);
    }
    if (WAVES == 4)
      asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
    else
      asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    __builtin_amdgcn_sched_barrier(0);
    __builtin_amdgcn_s_barrier();
    const bf16* Asp = Asb[t & 3];
    const bf16* Bsp = Bsb[t & 3];
    bf16x8 afrag[MF], bfrag[4];
#pragma unroll
    for (int f = 0; f < 4; ++f) {
      int rb = wn * 64 + f * 16 + lrow;
      bfrag[f] = *reinterpret_cast<const bf16x8*>(
          Bsp + rb * BK + (swz_chunk(rb, lk8 >> 3) << 3));
    }
#pragma unroll
    for (int f = 0; f < MF; ++f) {
      int ra = wm * MROWS + f * 16 + lrow;
      afrag[f] = *reinterpret_cast<const bf16x8*>(
          Asp + ra * BK + (swz_chunk(ra, lk8 >> 3) << 3));
    }
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int fm = 0; fm < MF; ++fm)
#pragma unroll
      for (int fn = 0; fn < 4; ++fn)
        acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afrag[fm], bfrag[fn], acc[fm][fn], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
  }

  int crow0 = tile_m + wm * MROWS + ((lane >> 4) << 2);
  int ccol0 = tile_n + wn * 64 + (lane & 15);
#pragma unroll
  for (int fm = 0; fm < MF; ++fm) {
#pragma unroll
    for (int fn = 0; fn < 4; ++fn) {
      int col = ccol0 + fn * 16;
      if (col >= N) continue;
      float badd = (bias != nullptr) ? bias[col] : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = crow0 + fm * 16 + r;
        if (row >= M) continue;
        float v = acc[fm][fn][r] + badd;
        if (relu && v < 0.f) v = 0.f;
        int64_t off = (int64_t)row * ldc + col;
        if (accum) v += bf2f(C[off]);
        C[off] = f2bf(v);
      }
    }
  }
}

// dw trans/trans GEMM with the col operand gathered implicitly:
// dw[Kg][Kcol] = dy^T @ im2col(x), optional fused db, split-K atomics
// (STORE 2) or plain store (STORE 1, single split).
template <int STORE>
__global__ __launch_bounds__(WNT, 1) void gemm_conv_dw_kernel(
    const bf16* __restrict__ A, const bf16* __restrict__ X,
    float* __restrict__ C, int M, int N, int K, int lda, int ldc,
    int ksplit, float alpha, float* __restrict__ db, CGeom gm) {
  __shared__ bf16 As[BM * BK];
  __shared__ bf16 Bs[WBN * BK];

  int mblocks = (M + BM - 1) / BM;
  int nblocks = (N + WBN - 1) / WBN;
  int bid = xcd_swizzle(blockIdx.x, mblocks * nblocks);
  int bm = bid / nblocks, bn = bid % nblocks;
  int tile_m = bm * BM, tile_n = bn * WBN;
  int k_begin = blockIdx.y * ksplit;
  int k_end = min(K, k_begin + ksplit);

  int tid = threadIdx.x;
  int wave = tid >> 6, lane = tid & 63;
  int wm = wave >> 2, wn = wave & 3;
  int lrow = lane & 15, lk8 = (lane >> 4) * 8;

  f32x4 acc[4][4] = {};
  bool do_db = (db != nullptr) && (bn == 0);
  for (int k0 = k_begin; k0 < k_end; k0 += BK) {
    stage_trans_pair_guarded(As, A, tile_m, M, lda, k0, k_end, tid, WNT,
                             BM);
    stage_trans_pair_implicit(Bs, X, gm, tile_n, k0, k_end, tid, WNT,
                              WBN);
    __syncthreads();
    if (do_db && tid < BM && tile_m + tid < M) {
      float acc_b = 0.f;
      const bf16* row = As + tid * BK;
#pragma unroll
      for (int kk = 0; kk < BK; ++kk)
        acc_b += bf2f(row[(swz_chunk(tid, kk >> 3) << 3) | (kk & 7)]);
      atomicAdd(db + tile_m + tid, acc_b);
    }
    bf16x8 afrag[4], bfrag[4];
#pragma unroll
    for (int f = 0; f < 4; ++f) {
      int ra = wm * 64 + f * 16 + lrow;
      int rb = wn * 64 + f * 16 + lrow;
      afrag[f] = *reinterpret_cast<const bf16x8*>(
          As + ra * BK + (swz_chunk(ra, lk8 >> 3) << 3));
      bfrag[f] = *reinterpret_cast<const bf16x8*>(
          Bs + rb * BK + (swz_chunk(rb, lk8 >> 3) << 3));
    }
#pragma unroll
    for (int fm = 0; fm < 4; ++fm)
#pragma unroll
      for (int fn = 0; fn < 4; ++fn)
        acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afrag[fm], bfrag[fn], acc[fm][fn], 0, 0, 0);
    __syncthreads();
  }

  int crow0 = tile_m + wm * 64 + ((lane >> 4) << 2);
  int ccol0 = tile_n + wn * 64 + (lane & 15);
#pragma unroll
  for (int fm = 0; fm < 4; ++fm)
#pragma unroll
    for (int fn = 0; fn < 4; ++fn) {
      int col = ccol0 + fn * 16;
      if (col >= N) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = crow0 + fm * 16 + r;
        if (row >= M) continue;
        if (STORE == 1)
          C[(int64_t)row * ldc + col] = acc[fm][fn][r] * alpha;
        else
          atomicAdd(C + (int64_t)row * ldc + col, acc[fm][fn][r] * alpha);
      }
    }
}

// ---- tr16 semantics probe: fills LDS with elem index, every lane does
// one ds_read_b64_tr_b16 at its own address and dumps v[0..3] — verifies
// the (addr + j*16) element pattern on hardware.
typedef __bf16 bf16x4 __attribute__((ext_vector_type(4)));

__global__ void tr16_probe_kernel(float* out, int mode) {
  __shared__ bf16 lds[2048];
  for (int i = threadIdx.x; i < 2048; i += blockDim.x)
    lds[i] = f2bf((float)i);
  __syncthreads();
  int lane = threadIdx.x & 63;
  int base;
  if (mode == 0) base = lane;                       // low bits sweep
  else if (mode == 1) base = lane * 16;             // tile sweep
  else if (mode == 2) base = (lane & 3) + (lane >> 2) * 16;  // quad cols
  else base = (lane & 15) + (lane >> 4) * 128;      // frag pattern
  auto* p = (__attribute__((address_space(3))) bf16x4*)(lds + base);
  bf16x4 v = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p);
#pragma unroll
  for (int j = 0; j < 4; ++j) out[lane * 4 + j] = bf2f(v[j]);
}

void tr16_probe(float* out, int mode, hipStream_t stream) {
  tr16_probe_kernel<<<1, 64, 0, stream>>>(out, mode);
}

// ------------------------------------------------------------------- host

void gemm_bf16_batched(const void* A_, const void* B_, void* C,
                       const float* bias, int M, int N, int K, int lda,
                       int ldb, int ldc, bool trans_a, bool trans_b,
                       int store_mode, int splitk, bool relu, float alpha,
                       int m_alloc, int n_alloc, int batch, int64_t sA,
                       int64_t sB, int64_t sC_bytes, hipStream_t stream) {
  if (m_alloc < M) m_alloc = M;
  if (n_alloc < N) n_alloc = N;
  const bf16* A = reinterpret_cast<const bf16*>(A_);
  const bf16* B = reinterpret_cast<const bf16*>(B_);
  int mblocks = (M + BM - 1) / BM, nblocks = (N + BN - 1) / BN;
  splitk = max(1, splitk);
  int ksplit = (K + splitk - 1) / splitk;
  ksplit = ((ksplit + BK - 1) / BK) * BK;  // keep split boundaries BK-aligned
  int zblocks = (K + ksplit - 1) / ksplit;
  if (zblocks > 1 && store_mode != 2)
    throw std::runtime_error("gemm: split-K requires atomic store mode");
  // store_mode 3 = bf16 RMW accumulate (single split only)
  dim3 grid(mblocks * nblocks, zblocks, batch);
  dim3 block(NTHREADS);

  // 8-wave blocks win on skinny shapes (one of M/N <= 256: fc6 +12%,
  // conv2/conv5 fwd +6-7%, conv1 +6% measured) and lose on fat ones
  // (conv3 -15%); COS_GEMM_W8=0/1 forces either variant for A/B.
  static const int w8_env = [] {
    const char* e = getenv("COS_GEMM_W8");
    return e ? (e[0] == '1' ? 1 : 0) : -1;
  }();
  bool w8 = (w8_env == -1) ? (M <= 256 || N <= 256) : (w8_env == 1);

#define COS_GEMM_CASE(TA, TB, SM)                                          \
  gemm_kernel<TA, TB, SM><<<grid, block, 0, stream>>>(                      \
      A, B, C, bias, M, N, K, lda, ldb, ldc, ksplit, relu ? 1 : 0, alpha,    \
      m_alloc, n_alloc, sA, sB, sC_bytes)

#define COS_GEMM_CASE8(SM)                                                 \
  gemm_kernel<false, false, SM, 8><<<grid, dim3(512), 0, stream>>>(        \
      A, B, C, bias, M, N, K, lda, ldb, ldc, ksplit, relu ? 1 : 0, alpha,   \
      m_alloc, n_alloc, sA, sB, sC_bytes)

#define COS_GEMM_SM(TA, TB)                                                 \
  do {                                                                      \
    if (store_mode == 0)      COS_GEMM_CASE(TA, TB, 0);                     \
    else if (store_mode == 1) COS_GEMM_CASE(TA, TB, 1);                     \
    else if (store_mode == 3) COS_GEMM_CASE(TA, TB, 3);                     \
    else                      COS_GEMM_CASE(TA, TB, 2);                     \
  } while (0)

  if (trans_a && trans_b && store_mode >= 1 && N > 128 && batch == 1) {
    int nb_w = (N + WBN - 1) / WBN;
    dim3 gw(mblocks * nb_w, zblocks);
    if (store_mode == 1)   // single-split: exclusive tiles, plain store
      gemm_tt_wide_kernel<1><<<gw, WNT, 0, stream>>>(
          A, B, reinterpret_cast<float*>(C), M, N, K, lda, ldb, ldc,
          ksplit, alpha, const_cast<float*>(bias));
    else
      gemm_tt_wide_kernel<2><<<gw, WNT, 0, stream>>>(
          A, B, reinterpret_cast<float*>(C), M, N, K, lda, ldb, ldc,
          ksplit, alpha, const_cast<float*>(bias));   // bias = fused db
    return;
  }
  if (!trans_a && !trans_b) {
    if (w8) {
      if (store_mode == 0)      COS_GEMM_CASE8(0);
      else if (store_mode == 1) COS_GEMM_CASE8(1);
      else if (store_mode == 3) COS_GEMM_CASE8(3);
      else                      COS_GEMM_CASE8(2);
      return;
    }
    COS_GEMM_SM(false, false);
  }
  else if (!trans_a && trans_b)  COS_GEMM_SM(false, true);
  else if (trans_a && !trans_b)  COS_GEMM_SM(true, false);
  else                           COS_GEMM_SM(true, true);
#undef COS_GEMM_SM
#undef COS_GEMM_CASE
#undef COS_GEMM_CASE8
}

static CGeom make_geom(const int* g) {
  // g = [H, W, C, P, Q, sh, sw, ph, pw, dil, S, c0, Cg, Kcol]
  CGeom gm;
  gm.H = g[0]; gm.W = g[1]; gm.C = g[2];
  gm.Q = g[4]; gm.PQ = g[3] * g[4];
  gm.sh = g[5]; gm.sw = g[6]; gm.ph = g[7]; gm.pw = g[8];
  gm.dil = g[9]; gm.S = g[10];
  gm.c0 = g[11]; gm.Cg = g[12]; gm.Kcol = g[13];
  gm.mPQ = magic40(gm.PQ);
  gm.mQ = magic40(gm.Q);
  gm.mCg = magic40(gm.Cg);
  gm.mS = magic40(gm.S);
  return gm;
}

void gemm_conv_fwd(const void* X, const void* B, void* C,
                   const float* bias, const void* zpage, int M, int N,
                   int K, int ldb, int ldc, bool relu, bool accum,
                   const int* geom, hipStream_t stream) {
  CGeom gm = make_geom(geom);
  int mblocks = (M + BM - 1) / BM, nblocks = (N + BN - 1) / BN;
  dim3 grid(mblocks * nblocks);
  static const int w8_env = [] {
    const char* e = getenv("COS_GEMM_W8");
    return e ? (e[0] == '1' ? 1 : 0) : -1;
  }();
  bool w8 = (w8_env == -1) ? (M <= 256 || N <= 256) : (w8_env == 1);
  static const int quad = [] {
    // COS_CONVQ: 0 = off, otherwise a max-K gate (1 = "all shapes").
    // Default 1152: same-box A/B is +2.2% CIFAR, neutral AlexNet/
    // GoogLeNet/LRCN; ungated it was -3% AlexNet (long-K shapes).
    const char* e = getenv("COS_CONVQ");
    if (!e) return 1152;
    int v = atoi(e);
    return v == 1 ? 1 << 30 : v;
  }();
  if (quad && w8 && K >= 4 * BK && K <= quad) {
    gemm_conv_fwd_q_kernel<8><<<grid, dim3(512), 0, stream>>>(
        (const bf16*)X, (const bf16*)B, (bf16*)C, bias,
        (const bf16*)zpage, M, N, K, ldb, ldc, relu ? 1 : 0,
        accum ? 1 : 0, gm);
    return;
  }
  if (w8)
    gemm_conv_fwd_kernel<8><<<grid, dim3(512), 0, stream>>>(
        (const bf16*)X, (const bf16*)B, (bf16*)C, bias,
        (const bf16*)zpage, M, N, K, ldb, ldc, relu ? 1 : 0,
        accum ? 1 : 0, gm);
  else
    gemm_conv_fwd_kernel<4><<<grid, dim3(256), 0, stream>>>(
        (const bf16*)X, (const bf16*)B, (bf16*)C, bias,
        (const bf16*)zpage, M, N, K, ldb, ldc, relu ? 1 : 0,
        accum ? 1 : 0, gm);
}

void gemm_conv_fwd_sc(const void* X, const void* B, void* C,
                      const float* bias, int M, int N, int K, int ldb,
                      int ldc, bool relu, const int* geom,
                      hipStream_t stream) {
  CGeom gm = make_geom(geom);
  int mblocks = (M + BM - 1) / BM, nblocks = (N + BN - 1) / BN;
  dim3 grid(mblocks * nblocks);
  static const int w8_env = [] {
    const char* e = getenv("COS_GEMM_W8");
    return e ? (e[0] == '1' ? 1 : 0) : -1;
  }();
  bool w8 = (w8_env == -1) ? (M <= 256 || N <= 256) : (w8_env == 1);
  if (w8)
    gemm_conv_fwd_sc_kernel<8><<<grid, dim3(512), 0, stream>>>(
        (const bf16*)X, (const bf16*)B, (bf16*)C, bias, M, N, K, ldb,
        ldc, relu ? 1 : 0, gm);
  else
    gemm_conv_fwd_sc_kernel<4><<<grid, dim3(256), 0, stream>>>(
        (const bf16*)X, (const bf16*)B, (bf16*)C, bias, M, N, K, ldb,
        ldc, relu ? 1 : 0, gm);
}

void gemm_conv_dw_sc(const void* A, const void* X, float* C, int M,
                     int N, int K, int lda, int ldc, int store_mode,
                     int splitk, float alpha, float* db, const int* geom,
                     hipStream_t stream) {
  CGeom gm = make_geom(geom);
  splitk = max(1, splitk);
  int ksplit = (K + splitk - 1) / splitk;
  ksplit = ((ksplit + BK - 1) / BK) * BK;
  int zblocks = (K + ksplit - 1) / ksplit;
  int mblocks = (M + BM - 1) / BM, nblocks = (N + WBN - 1) / WBN;
  dim3 grid(mblocks * nblocks, zblocks);
  if (store_mode == 1 && zblocks == 1)
    gemm_conv_dw_sc_kernel<1><<<grid, WNT, 0, stream>>>(
        (const bf16*)A, (const bf16*)X, C, M, N, K, lda, ldc, ksplit,
        alpha, db, gm);
  else
    gemm_conv_dw_sc_kernel<2><<<grid, WNT, 0, stream>>>(
        (const bf16*)A, (const bf16*)X, C, M, N, K, lda, ldc, ksplit,
        alpha, db, gm);
}

void gemm_conv_dw(const void* A, const void* X, float* C, int M, int N,
                  int K, int lda, int ldc, int store_mode, int splitk,
                  float alpha, float* db, const int* geom,
                  hipStream_t stream) {
  CGeom gm = make_geom(geom);
  splitk = max(1, splitk);
  int ksplit = (K + splitk - 1) / splitk;
  ksplit = ((ksplit + BK - 1) / BK) * BK;
  int zblocks = (K + ksplit - 1) / ksplit;
  int mblocks = (M + BM - 1) / BM, nblocks = (N + WBN - 1) / WBN;
  dim3 grid(mblocks * nblocks, zblocks);
  if (store_mode == 1 && zblocks == 1)
    gemm_conv_dw_kernel<1><<<grid, WNT, 0, stream>>>(
        (const bf16*)A, (const bf16*)X, C, M, N, K, lda, ldc, ksplit,
        alpha, db, gm);
  else
    gemm_conv_dw_kernel<2><<<grid, WNT, 0, stream>>>(
        (const bf16*)A, (const bf16*)X, C, M, N, K, lda, ldc, ksplit,
        alpha, db, gm);
}

void gemm_bf16(const void* A_, const void* B_, void* C, const float* bias,
               int M, int N, int K, int lda, int ldb, int ldc,
               bool trans_a, bool trans_b, int store_mode, int splitk,
               bool relu, float alpha, int m_alloc, int n_alloc,
               hipStream_t stream) {
  gemm_bf16_batched(A_, B_, C, bias, M, N, K, lda, ldb, ldc, trans_a,
                    trans_b, store_mode, splitk, relu, alpha, m_alloc,
                    n_alloc, 1, 0, 0, 0, stream);
}

}  // namespace cosamd
