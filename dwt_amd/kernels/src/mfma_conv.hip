// MFMA implicit-GEMM convolution / GEMM for MI355X (gfx950), bf16 NHWC.
//
// One kernel covers the R50 backbone's forward conv shapes:
//   out[n,p,q,co] = sum_{r,s,ci} x[n, p*stride-pad+r, q*stride-pad+s, ci]
//                               * w[co, r, s, ci]            (+bias, +ReLU)
// as the GEMM  C[M,N] = A[M,K] @ Bt[N,K]^T with
//   M = N*P*Q output positions, N = Cout, K = KH*KW*Cin,
//   A rows gathered from the NHWC input on the fly (im2col-free),
//   Bt = the channels_last conv weight as stored ([co][r][s][ci]) — no
//   weight reshape needed.  KH=KW=1, stride=1, pad=0 degenerates to a plain
//   GEMM (fc layers, 1x1 convs).
//
// Structure (cdna_hip_programming.md §5 canonical anatomy, reg-staged):
//   128x128 block tile, BK=64, 4 waves each computing a 64x64 sub-tile as
//   4x4 fragments of v_mfma_f32_16x16x32_bf16; A/B tiles staged via
//   registers into LDS with +16B row padding (bank-conflict fix, §6 G4);
//   fp32 accumulate; fused bias + ReLU epilogue, bf16 store.
//
// A-fragment layout for mfma_f32_16x16x32_bf16 (cdna4_isa.md §10):
//   lane l holds A[row = l%16][k = (l/16)*8 + j], j = 0..7  (one b128 read)
//   B operand: lane l holds B[k = (l/16)*8 + j][col = l%16], which equals
//   Bt[col][k] — so Bt rows load with the SAME pattern as A rows.
//   C/D: lane l, reg r -> row = (l/16)*4 + r, col = l%16.
// Verified transpose-safe on hardware by tests/test_gpu_mfma.py against
// torch.matmul / MIOpen conv on asymmetric inputs.

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include <hip/hip_runtime.h>

#include <cstdint>

namespace dwtmm {

using bf16 = c10::BFloat16;
typedef __attribute__((ext_vector_type(8))) short short8;   // bf16 x8 frag
typedef __attribute__((ext_vector_type(4))) float floatx4;  // fp32 x4 acc

#define DEV_INLINE __device__ __forceinline__

constexpr int BM = 128, BN = 128, BK = 64;
constexpr int THREADS = 256;                // 4 waves: 2x2 of 64x64
constexpr int PAD_HALFS = 8;                // +16 B per LDS row
constexpr int LDS_PITCH = BK + PAD_HALFS;   // halves (bf16 units)

struct ConvParams {
  int N, H, W, Cin;      // input
  int P, Q, Cout;        // output spatial + channels
  int KH, KW, stride, pad;
  int64_t M;             // N*P*Q (fwd) / N*H*W (dgrad)
  int K;                 // KH*KW*Cin (fwd) / KH*KW*Cdy (dgrad)
  int Cdy;               // dgrad: channels of dy
  int cShift = -1;       // log2 of the K-fastest channel dim if pow2
};

DEV_INLINE float bf16_to_f(unsigned short u) {
  union { unsigned int i; float f; } v;
  v.i = (unsigned int)u << 16;
  return v.f;
}
DEV_INLINE unsigned short f_to_bf16(float f) {
  union { unsigned int i; float f2; } v;
  v.f2 = f;
  unsigned int lsb = (v.i >> 16) & 1u;
  return (unsigned short)((v.i + 0x7fffu + lsb) >> 16);
}

// Tile staging, split into LOAD (issue global reads into registers, early)
// and WRITE (LDS store, late) so HBM latency hides under the MFMA phase
// (guide §6 G15 async-STAGE split / T14).  Each thread moves 32 halves as 4
// chunks of 8; a chunk-of-8 stays within one (r,s) patch element when
// Cin % 8 == 0.
struct StageRegs {
  uint4 c[4];
};

// A-operand gather modes
enum AMode { A_DENSE = 0, A_CONV = 1, A_DGRAD = 2 };

// Per-tile row cache: each thread's 4 chunk rows keep the same output
// position m across the whole K loop, so the m -> (n, p, q) decode (two
// 64-bit div/mod chains) runs ONCE per tile instead of once per k-step.
struct RowCache {
  int h0[4], w0[4];     // A_CONV: p*stride-pad / q*stride-pad; A_DGRAD: h+pad
  int64_t base[4];      // image base offset (elements)
  bool valid[4];
};

template <int MODE>
DEV_INLINE RowCache make_rows(const ConvParams& cp, int64_t m0) {
  RowCache rc;
  const int t = threadIdx.x;
#pragma unroll
  for (int c = 0; c < 4; ++c) {
    const int idx = t + c * THREADS;
    const int row = idx / (BK / 8);
    const int64_t m = m0 + row;
    rc.valid[c] = m < cp.M;
    const int64_t mm = rc.valid[c] ? m : 0;
    if (MODE == A_CONV) {
      const int q = (int)(mm % cp.Q);
      const int64_t np = mm / cp.Q;
      const int p = (int)(np % cp.P);
      const int n = (int)(np / cp.P);
      rc.h0[c] = p * cp.stride - cp.pad;
      rc.w0[c] = q * cp.stride - cp.pad;
      rc.base[c] = (int64_t)n * cp.H * cp.W * cp.Cin;
    } else if (MODE == A_DGRAD) {
      const int wi = (int)(mm % cp.W);
      const int64_t nh = mm / cp.W;
      const int hi = (int)(nh % cp.H);
      const int n = (int)(nh / cp.H);
      rc.h0[c] = hi + cp.pad;
      rc.w0[c] = wi + cp.pad;
      rc.base[c] = (int64_t)n * cp.P * cp.Q * cp.Cdy;
    } else {
      rc.h0[c] = 0; rc.w0[c] = 0;
      rc.base[c] = mm * cp.K;
    }
  }
  return rc;
}

// cShift >= 0 when the K-fastest channel count (Cin fwd / Cdy dgrad) is a
// power of two: replaces the per-chunk div/mod with shift/mask.
DEV_INLINE void split_kc(int kg, int cdim, int cshift, int& rs, int& cc) {
  if (cshift >= 0) {
    cc = kg & (cdim - 1);
    rs = kg >> cshift;
  } else {
    cc = kg % cdim;
    rs = kg / cdim;
  }
}

template <int MODE>
DEV_INLINE void load_a(const bf16* __restrict__ x, const ConvParams& cp,
                       const RowCache& rc, int k0, StageRegs& rg) {
  const int t = threadIdx.x;
#pragma unroll
  for (int c = 0; c < 4; ++c) {
    const int idx = t + c * THREADS;          // chunk index in tile
    const int kc = (idx % (BK / 8)) * 8;      // chunk k offset in tile
    const int kg = k0 + kc;
    if (!rc.valid[c] || kg >= cp.K) {
      rg.c[c] = make_uint4(0, 0, 0, 0);
      continue;
    }
    if (MODE == A_DENSE) {
      rg.c[c] = *reinterpret_cast<const uint4*>(x + rc.base[c] + kg);
      continue;
    }
    if (MODE == A_DGRAD) {
      // x here is dy (N,P,Q,Cdy) raw NHWC; k -> (r, s, co), co fastest.
      // dx[n,h,w,ci] needs dy[n, (h+pad-r)/stride, (w+pad-s)/stride, co].
      int rs, co;
      split_kc(kg, cp.Cdy, cp.cShift, rs, co);
      const int sx = rs % cp.KW;
      const int r = rs / cp.KW;
      const int hp = rc.h0[c] - r;
      const int wp = rc.w0[c] - sx;
      bool ok = hp >= 0 && wp >= 0 && hp % cp.stride == 0 && wp % cp.stride == 0;
      const int pp = hp / cp.stride, qq = wp / cp.stride;
      ok = ok && pp < cp.P && qq < cp.Q;
      if (!ok) {
        rg.c[c] = make_uint4(0, 0, 0, 0);
      } else if (cp.Cdy % 8 == 0) {
        rg.c[c] = *reinterpret_cast<const uint4*>(
            x + rc.base[c] + ((int64_t)pp * cp.Q + qq) * cp.Cdy + co);
      } else {
        unsigned short tmp[8];
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const int k = kg + j;
          unsigned short v = 0;
          if (k < cp.K && k / cp.Cdy == rs)
            v = *reinterpret_cast<const unsigned short*>(
                x + rc.base[c] + ((int64_t)pp * cp.Q + qq) * cp.Cdy + k % cp.Cdy);
          tmp[j] = v;
        }
        rg.c[c] = *reinterpret_cast<const uint4*>(tmp);
      }
      continue;
    }
    if (cp.Cin % 8 == 0) {
      int rs, ci;
      split_kc(kg, cp.Cin, cp.cShift, rs, ci);
      const int s = rs % cp.KW;
      const int r = rs / cp.KW;
      const int h = rc.h0[c] + r;
      const int w = rc.w0[c] + s;
      if (h < 0 || h >= cp.H || w < 0 || w >= cp.W) {
        rg.c[c] = make_uint4(0, 0, 0, 0);
      } else {
        rg.c[c] = *reinterpret_cast<const uint4*>(
            x + rc.base[c] + ((int64_t)h * cp.W + w) * cp.Cin + ci);
      }
    } else if (cp.Cin == 4) {
      // stem path (3-channel input zero-padded to 4 on the host): a chunk
      // of 8 halves = two horizontally adjacent pixels; when both are in
      // bounds on the same kernel row it is ONE 16-B load
      const int rs0 = kg >> 2;          // kg is 8-aligned -> ci = 0
      const int rs1 = rs0 + 1;
      const int s0 = rs0 % cp.KW, r0 = rs0 / cp.KW;
      const int h0 = rc.h0[c] + r0, w0 = rc.w0[c] + s0;
      const bool in0 = h0 >= 0 && h0 < cp.H && w0 >= 0 && w0 < cp.W;
      if (in0 && s0 + 1 < cp.KW && w0 + 1 < cp.W && kg + 4 < cp.K) {
        rg.c[c] = *reinterpret_cast<const uint4*>(
            x + rc.base[c] + ((int64_t)h0 * cp.W + w0) * 4);
      } else {
        uint2 lo = make_uint2(0, 0), hi = make_uint2(0, 0);
        if (in0)
          lo = *reinterpret_cast<const uint2*>(
              x + rc.base[c] + ((int64_t)h0 * cp.W + w0) * 4);
        if (kg + 4 < cp.K) {
          const int s1 = rs1 % cp.KW, r1 = rs1 / cp.KW;
          const int h1 = rc.h0[c] + r1, w1 = rc.w0[c] + s1;
          if (h1 >= 0 && h1 < cp.H && w1 >= 0 && w1 < cp.W)
            hi = *reinterpret_cast<const uint2*>(
                x + rc.base[c] + ((int64_t)h1 * cp.W + w1) * 4);
        }
        rg.c[c] = make_uint4(lo.x, lo.y, hi.x, hi.y);
      }
    } else {
      unsigned short tmp[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int k = kg + j;
        unsigned short v = 0;
        if (k < cp.K) {
          const int ci = k % cp.Cin;
          const int rs = k / cp.Cin;
          const int s = rs % cp.KW;
          const int r = rs / cp.KW;
          const int h = rc.h0[c] + r;
          const int w = rc.w0[c] + s;
          if (h >= 0 && h < cp.H && w >= 0 && w < cp.W)
            v = *reinterpret_cast<const unsigned short*>(
                x + rc.base[c] + ((int64_t)h * cp.W + w) * cp.Cin + ci);
        }
        tmp[j] = v;
      }
      rg.c[c] = *reinterpret_cast<const uint4*>(tmp);
    }
  }
}

DEV_INLINE void load_b(const bf16* __restrict__ wgt, int ncols, int K,
                       int n0, int k0, StageRegs& rg) {
  const int t = threadIdx.x;
#pragma unroll
  for (int c = 0; c < 4; ++c) {
    const int idx = t + c * THREADS;
    const int row = idx / (BK / 8);
    const int kc = (idx % (BK / 8)) * 8;
    const int n = n0 + row;
    const int kg = k0 + kc;
    if (n >= ncols || kg >= K) {
      rg.c[c] = make_uint4(0, 0, 0, 0);
    } else if (kg + 8 <= K) {
      rg.c[c] = *reinterpret_cast<const uint4*>(wgt + (int64_t)n * K + kg);
    } else {
      unsigned short tmp[8];
#pragma unroll
      for (int j = 0; j < 8; ++j)
        tmp[j] = (kg + j < K)
            ? *reinterpret_cast<const unsigned short*>(wgt + (int64_t)n * K + kg + j)
            : (unsigned short)0;
      rg.c[c] = *reinterpret_cast<const uint4*>(tmp);
    }
  }
}

DEV_INLINE void write_tile(const StageRegs& rg, unsigned short* lds) {
  const int t = threadIdx.x;
#pragma unroll
  for (int c = 0; c < 4; ++c) {
    const int idx = t + c * THREADS;
    const int row = idx / (BK / 8);
    const int kc = (idx % (BK / 8)) * 8;
    *reinterpret_cast<uint4*>(lds + row * LDS_PITCH + kc) = rg.c[c];
  }
}

// PIPE: double-buffered LDS + issue-early/write-late staging (wins for
// long-K dense GEMMs, measured +24-34% at K>=2048); the simple
// single-buffer loop wins for short-K and for the implicit gather path
// (within-shape A/B on the R50 shapes).  SWZ: XCD-aware bijective block
// remap (guide T1) — only when the grid has several N-tiles to share.
template <int MODE, bool RELU, bool HAS_BIAS, bool PIPE, bool SWZ>
__global__ __launch_bounds__(THREADS, PIPE ? 2 : 3) void conv_implicit_gemm_kernel(
    const bf16* __restrict__ x, const bf16* __restrict__ wgt,
    const float* __restrict__ bias, bf16* __restrict__ out, ConvParams cp,
    int mtiles, int ntiles) {
  __shared__ unsigned short lds_a[PIPE ? 2 : 1][BM * LDS_PITCH];
  __shared__ unsigned short lds_b[PIPE ? 2 : 1][BN * LDS_PITCH];

  int bid = blockIdx.x;
  if (SWZ) {
    const int nwg = mtiles * ntiles;
    const int nx = 8;
    const int qq = nwg / nx, rr = nwg % nx;
    const int xcd = bid % nx, idx = bid / nx;
    bid = (xcd < rr ? xcd * (qq + 1) : rr * (qq + 1) + (xcd - rr) * qq) + idx;
  }
  const int64_t m0 = (int64_t)(bid / ntiles) * BM;
  const int n0 = (bid % ntiles) * BN;

  const int wave = threadIdx.x / 64;
  const int lane = threadIdx.x % 64;
  const int wm = (wave / 2) * 64;   // wave's 64x64 sub-tile origin
  const int wn = (wave % 2) * 64;

  floatx4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = (floatx4){0.f, 0.f, 0.f, 0.f};

  const int frag_row = lane % 16;
  const int frag_koff = (lane / 16) * 8;
  const int nk = (cp.K + BK - 1) / BK;

  StageRegs ra, rb;
  const RowCache rc = make_rows<MODE>(cp, m0);
  if (PIPE) {
    load_a<MODE>(x, cp, rc, 0, ra);
    load_b(wgt, cp.Cout, cp.K, n0, 0, rb);
    write_tile(ra, lds_a[0]);
    write_tile(rb, lds_b[0]);
    __syncthreads();
  }

  for (int t = 0; t < nk; ++t) {
    const int cur = PIPE ? (t & 1) : 0;
    if (PIPE) {
      // issue next tile's global loads now — they stay in flight under the
      // MFMA phase and are only waited for at the ds_write below
      if (t + 1 < nk) {
        load_a<MODE>(x, cp, rc, (t + 1) * BK, ra);
        load_b(wgt, cp.Cout, cp.K, n0, (t + 1) * BK, rb);
      }
    } else {
      load_a<MODE>(x, cp, rc, t * BK, ra);
      load_b(wgt, cp.Cout, cp.K, n0, t * BK, rb);
      write_tile(ra, lds_a[0]);
      write_tile(rb, lds_b[0]);
      __syncthreads();
    }
#pragma unroll
    for (int ks = 0; ks < BK; ks += 32) {
      short8 afrag[4], bfrag[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const unsigned short* pa =
            lds_a[cur] + (wm + i * 16 + frag_row) * LDS_PITCH + ks + frag_koff;
        afrag[i] = *reinterpret_cast<const short8*>(pa);
        const unsigned short* pb =
            lds_b[cur] + (wn + i * 16 + frag_row) * LDS_PITCH + ks + frag_koff;
        bfrag[i] = *reinterpret_cast<const short8*>(pb);
      }
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[i], bfrag[j], acc[i][j], 0, 0, 0);
    }
    if (PIPE) {
      if (t + 1 < nk) {
        __syncthreads();  // everyone finished reading buf[cur^1] (tile t-1)
        write_tile(ra, lds_a[cur ^ 1]);
        write_tile(rb, lds_b[cur ^ 1]);
        __syncthreads();
      }
    } else {
      __syncthreads();
    }
  }

  // epilogue: lane l, reg r -> row (l/16)*4 + r, col l%16 of each 16x16 frag
  const int erow = (lane / 16) * 4;
  const int ecol = lane % 16;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int64_t m = m0 + wm + i * 16 + erow + r;
      if (m >= cp.M) continue;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int n = n0 + wn + j * 16 + ecol;
        if (n >= cp.Cout) continue;
        float v = acc[i][j][r];
        if (HAS_BIAS) v += bias[n];
        if (RELU) v = fmaxf(v, 0.f);
        *reinterpret_cast<unsigned short*>(out + m * cp.Cout + n) = f_to_bf16(v);
      }
    }
  }
}

// ===========================================================================
// wgrad v2: dw[co][(r,s,ci)] = sum_{k=npq} dy[k][co] * xpatch[k][(r,s,ci)]
//
// The K dimension is the (huge) output-position axis, so the kernel is
// SPLIT-K: the grid is (mtiles * ntiles * splitk) and each block accumulates
// its K-slab into an fp32 workspace with atomicAdd; a tiny convert kernel
// writes the bf16 dw afterwards.  Both operands are K-major with stride C
// in memory, so tiles are loaded coalesced in (k, c) orientation and
// TRANSPOSED through LDS into the (c, k) fragment layout the MFMA wants.
// 64x64 tile, BK=64, 4 waves each doing a 32x32 quadrant.
//
// B (x patches): when Cin % 64 == 0 every 64-column N-tile lies inside one
// (r, s) tap, so the gather is one shifted NHWC base + coalesced 16-B
// chunks; otherwise (stem Cin=3) a scalar per-element gather.
// ===========================================================================



constexpr int WBK = 64;
constexpr int WPITCH = WBK;  // no pad: the chunk XOR swizzle below spreads
                             // banks instead

// LDS transpose swizzle: element (row, k) lives at
//   row*WPITCH + ((k>>3) ^ wswz(row))*8 + (k&7)
// wswz varies with BOTH row%8 and row/8, so the transpose's strided stores
// (8 rows x one k-pair per wave instruction) spread across all banks
// (un-swizzled they land 16-way conflicted), while b128 fragment reads of
// 8-aligned k-chunks stay contiguous.
DEV_INLINE int wswz(int row) { return ((row >> 3) ^ row) & 7; }

// WT = square tile side (64 for small Cout/RSC shapes, 128 otherwise —
// the 64 tile re-reads operands across tiles and goes HBM-bound on big
// shapes; the 128 tile wastes MFMA work when M or N < 128).
template <bool ALIGNED_B, int WT>
__global__ __launch_bounds__(256, WT == 64 ? 4 : 3) void wgrad_splitk_kernel(
    const bf16* __restrict__ dy, const bf16* __restrict__ x,
    float* __restrict__ ws, ConvParams cp, int mtiles, int ntiles,
    int splitk, int64_t klen) {
  constexpr int NP = WT / 64;   // load passes per operand
  constexpr int NF = WT / 32;   // MFMA fragments per wave dim
  __shared__ unsigned short lds_a[WT * WPITCH];  // [co][k]
  __shared__ unsigned short lds_b[WT * WPITCH];  // [rsci][k]

  const int bid = blockIdx.x;
  const int tile = bid / splitk;
  const int kchunk = bid % splitk;
  const int m0 = (tile / ntiles) * WT;   // co origin
  const int n0 = (tile % ntiles) * WT;   // rsci origin
  const int64_t ks = (int64_t)kchunk * klen;
  const int64_t ke = (ks + klen < cp.M) ? ks + klen : cp.M;  // cp.M = real NPQ

  const int t = threadIdx.x;
  const int wave = t / 64;
  const int lane = t % 64;
  const int wm = (wave / 2) * (WT / 2);
  const int wn = (wave % 2) * (WT / 2);

  floatx4 acc[NF][NF];
#pragma unroll
  for (int i = 0; i < NF; ++i)
#pragma unroll
    for (int j = 0; j < NF; ++j) acc[i][j] = (floatx4){0.f, 0.f, 0.f, 0.f};

  const int frag_row = lane % 16;
  const int frag_koff = (lane / 16) * 8;

  // thread t owns channel chunk (t%8)*8 (two 64-row passes) and the k-pair
  // 2*(t/8): adjacent k values pack into ds_write_b32 (per-element u16
  // transpose stores cost 4x the LDS write cycles)
  const int cch = (t % 8) * 8;
  const int kp = 2 * (t / 8);

  for (int64_t k0 = ks; k0 < ke; k0 += WBK) {
    const int64_t ka = k0 + kp;
    // ---- dy tile [WBK k][WT co] -> lds_a[co][k] ----
#pragma unroll
    for (int pass = 0; pass < NP; ++pass) {
      const int c2 = cch + pass * 64;
      uint4 u0 = make_uint4(0, 0, 0, 0), u1 = make_uint4(0, 0, 0, 0);
      if (m0 + c2 < cp.N) {  // cp.N = Cout here
        if (ka < ke)
          u0 = *reinterpret_cast<const uint4*>(dy + ka * cp.N + m0 + c2);
        if (ka + 1 < ke)
          u1 = *reinterpret_cast<const uint4*>(dy + (ka + 1) * cp.N + m0 + c2);
      }
      const unsigned short* v0 = reinterpret_cast<const unsigned short*>(&u0);
      const unsigned short* v1 = reinterpret_cast<const unsigned short*>(&u1);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int row = c2 + j;
        const unsigned int pk = (unsigned int)v0[j] | ((unsigned int)v1[j] << 16);
        *reinterpret_cast<unsigned int*>(
            lds_a + row * WPITCH + (((kp >> 3) ^ wswz(row)) << 3) + (kp & 7)) = pk;
      }
    }
    // ---- x-patch tile [WBK k][WT rsci] -> lds_b[rsci][k] ----
#pragma unroll
    for (int pass = 0; pass < NP; ++pass) {
      const int c2 = cch + pass * 64;
      unsigned short v0[8], v1[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) { v0[j] = 0; v1[j] = 0; }
      // when Cin % 64 == 0 the 8-row chunk (64-aligned) sits in ONE tap
      int rB = 0, sB = 0, ciB = 0;
      if (ALIGNED_B) {
        const int rs = (n0 + c2) / cp.Cin;
        sB = rs % cp.KW;
        rB = rs / cp.KW;
        ciB = (n0 + c2) % cp.Cin;
      }
#pragma unroll
      for (int half = 0; half < 2; ++half) {
        const int64_t k = ka + half;
        unsigned short* v = half ? v1 : v0;
        if (k >= ke) continue;
        const int q = (int)(k % cp.Q);
        const int64_t np = k / cp.Q;
        const int p = (int)(np % cp.P);
        const int n = (int)(np / cp.P);
        if (ALIGNED_B) {
          const int h = p * cp.stride - cp.pad + rB;
          const int w = q * cp.stride - cp.pad + sB;
          if (h >= 0 && h < cp.H && w >= 0 && w < cp.W)
            *reinterpret_cast<uint4*>(v) = *reinterpret_cast<const uint4*>(
                x + (((int64_t)n * cp.H + h) * cp.W + w) * cp.Cin + ciB);
        } else {
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            const int nr = n0 + c2 + j;
            if (nr >= cp.Cout) break;   // cp.Cout = KH*KW*Cin columns
            const int ci = nr % cp.Cin;
            const int rs = nr / cp.Cin;
            const int sx = rs % cp.KW;
            const int r = rs / cp.KW;
            const int h = p * cp.stride - cp.pad + r;
            const int w = q * cp.stride - cp.pad + sx;
            if (h >= 0 && h < cp.H && w >= 0 && w < cp.W)
              v[j] = *reinterpret_cast<const unsigned short*>(
                  x + (((int64_t)n * cp.H + h) * cp.W + w) * cp.Cin + ci);
          }
        }
      }
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int row = c2 + j;
        const unsigned int pk = (unsigned int)v0[j] | ((unsigned int)v1[j] << 16);
        *reinterpret_cast<unsigned int*>(
            lds_b + row * WPITCH + (((kp >> 3) ^ wswz(row)) << 3) + (kp & 7)) = pk;
      }
    }
    __syncthreads();
#pragma unroll
    for (int kstep = 0; kstep < WBK; kstep += 32) {
      short8 afrag[NF], bfrag[NF];
      const int kb = (kstep + frag_koff) >> 3;  // 8-aligned chunk index
#pragma unroll
      for (int i = 0; i < NF; ++i) {
        const int ra = wm + i * 16 + frag_row;
        const int rb = wn + i * 16 + frag_row;
        afrag[i] = *reinterpret_cast<const short8*>(
            lds_a + ra * WPITCH + ((kb ^ wswz(ra)) << 3));
        bfrag[i] = *reinterpret_cast<const short8*>(
            lds_b + rb * WPITCH + ((kb ^ wswz(rb)) << 3));
      }
#pragma unroll
      for (int i = 0; i < NF; ++i)
#pragma unroll
        for (int j = 0; j < NF; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[i], bfrag[j], acc[i][j], 0, 0, 0);
    }
    __syncthreads();
  }

  // epilogue: atomicAdd fp32 workspace [Cout][RSC]
  const int erow = (lane / 16) * 4;
  const int ecol = lane % 16;
#pragma unroll
  for (int i = 0; i < NF; ++i) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int m = m0 + wm + i * 16 + erow + r;
      if (m >= cp.N) continue;
#pragma unroll
      for (int j = 0; j < NF; ++j) {
        const int n = n0 + wn + j * 16 + ecol;
        if (n >= cp.Cout) continue;
        atomicAdd(&ws[(int64_t)m * cp.Cout + n], acc[i][j][r]);
      }
    }
  }
}

__global__ void f32_to_bf16_kernel(const float* __restrict__ src,
                                   bf16* __restrict__ dst, int64_t n) {
  const int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n)
    *reinterpret_cast<unsigned short*>(dst + i) = f_to_bf16(src[i]);
}

// ===========================================================================
// Class-decomposed strided dgrad.  For stride s, dx positions split into s*s
// residue classes (h%s, w%s); each class has a FIXED subset of (r,s) taps
// that satisfy the stride-divisibility, so the per-class GEMM runs only the
// useful K (ntaps*Cdy) instead of wading through KH*KW*Cdy that is ~3/4
// zeros (the single-kernel gather measured 0.13-0.22x library on the R50
// stride-2 shapes).  One launch per class; 128x128 tile as the main kernel.
// ===========================================================================

struct DgradClsParams {
  int H, W, Cin, P, Q, Cdy, Kfull;  // Kfull = KH*KW*Cdy (wd row pitch)
  int ch, cw, stride;
  int Hc, Wc;              // class spatial extent
  int64_t M;               // N*Hc*Wc
  int K;                   // ntaps*Cdy
  int cShift;              // log2(Cdy), Cdy % 8 == 0 guaranteed
  int dp[16], dq[16], rsIdx[16];
};

__global__ __launch_bounds__(THREADS, 3) void dgrad_cls_kernel(
    const bf16* __restrict__ dy, const bf16* __restrict__ wd,
    bf16* __restrict__ dx, DgradClsParams gp, int mtiles, int ntiles) {
  __shared__ unsigned short lds_a[BM * LDS_PITCH];
  __shared__ unsigned short lds_b[BN * LDS_PITCH];
  const int bid = blockIdx.x;
  const int64_t m0 = (int64_t)(bid / ntiles) * BM;
  const int n0 = (bid % ntiles) * BN;
  const int wave = threadIdx.x / 64;
  const int lane = threadIdx.x % 64;
  const int wm = (wave / 2) * 64;
  const int wn = (wave % 2) * 64;

  floatx4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = (floatx4){0.f, 0.f, 0.f, 0.f};

  // row cache: class position m -> (n, hh, ww)
  int rcP[4], rcQ[4];
  int64_t rbase[4];
  bool rvalid[4];
  {
    const int t = threadIdx.x;
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      const int idx = t + c * THREADS;
      const int row = idx / (BK / 8);
      const int64_t m = m0 + row;
      rvalid[c] = m < gp.M;
      const int64_t mm = rvalid[c] ? m : 0;
      const int ww = (int)(mm % gp.Wc);
      const int64_t nh = mm / gp.Wc;
      const int hh = (int)(nh % gp.Hc);
      const int n = (int)(nh / gp.Hc);
      rcP[c] = hh;
      rcQ[c] = ww;
      rbase[c] = (int64_t)n * gp.P * gp.Q * gp.Cdy;
    }
  }

  const int frag_row = lane % 16;
  const int frag_koff = (lane / 16) * 8;
  const int nk = (gp.K + BK - 1) / BK;
  const int t = threadIdx.x;

  for (int kt = 0; kt < nk; ++kt) {
    const int k0 = kt * BK;
    // A: dy gather
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      const int idx = t + c * THREADS;
      const int kc = (idx % (BK / 8)) * 8;
      const int kg = k0 + kc;
      uint4 val = make_uint4(0, 0, 0, 0);
      if (rvalid[c] && kg < gp.K) {
        const int tap = kg >> gp.cShift;
        const int co = kg & (gp.Cdy - 1);
        const int pp = rcP[c] + gp.dp[tap];
        const int qq = rcQ[c] + gp.dq[tap];
        if (pp >= 0 && pp < gp.P && qq >= 0 && qq < gp.Q)
          val = *reinterpret_cast<const uint4*>(
              dy + rbase[c] + ((int64_t)pp * gp.Q + qq) * gp.Cdy + co);
      }
      const int row = (t + c * THREADS) / (BK / 8);
      *reinterpret_cast<uint4*>(lds_a + row * LDS_PITCH + kc) = val;
    }
    // B: wd rows (ci), k -> (tap, co) -> wd[n][rsIdx[tap]*Cdy + co]
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      const int idx = t + c * THREADS;
      const int row = idx / (BK / 8);
      const int kc = (idx % (BK / 8)) * 8;
      const int kg = k0 + kc;
      const int n = n0 + row;
      uint4 val = make_uint4(0, 0, 0, 0);
      if (n < gp.Cin && kg < gp.K) {
        const int tap = kg >> gp.cShift;
        const int co = kg & (gp.Cdy - 1);
        val = *reinterpret_cast<const uint4*>(
            wd + (int64_t)n * gp.Kfull + gp.rsIdx[tap] * gp.Cdy + co);
      }
      *reinterpret_cast<uint4*>(lds_b + row * LDS_PITCH + kc) = val;
    }
    __syncthreads();
#pragma unroll
    for (int ks = 0; ks < BK; ks += 32) {
      short8 afrag[4], bfrag[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        afrag[i] = *reinterpret_cast<const short8*>(
            lds_a + (wm + i * 16 + frag_row) * LDS_PITCH + ks + frag_koff);
        bfrag[i] = *reinterpret_cast<const short8*>(
            lds_b + (wn + i * 16 + frag_row) * LDS_PITCH + ks + frag_koff);
      }
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[i], bfrag[j], acc[i][j], 0, 0, 0);
    }
    __syncthreads();
  }

  // epilogue: scatter to dx[n, hh*s+ch, ww*s+cw, ci]
  const int erow = (lane / 16) * 4;
  const int ecol = lane % 16;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int64_t m = m0 + wm + i * 16 + erow + r;
      if (m >= gp.M) continue;
      const int ww = (int)(m % gp.Wc);
      const int64_t nh = m / gp.Wc;
      const int hh = (int)(nh % gp.Hc);
      const int n = (int)(nh / gp.Hc);
      const int h = hh * gp.stride + gp.ch;
      const int w = ww * gp.stride + gp.cw;
      const int64_t obase = (((int64_t)n * gp.H + h) * gp.W + w) * gp.Cin;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int col = n0 + wn + j * 16 + ecol;
        if (col >= gp.Cin) continue;
        *reinterpret_cast<unsigned short*>(dx + obase + col) =
            f_to_bf16(acc[i][j][r]);
      }
    }
  }
}

}  // namespace dwtmm

// ---------------------------------------------------------------------------
// launchers (referenced from dwt_kernels.hip bindings)
// ---------------------------------------------------------------------------

using torch::Tensor;

static inline hipStream_t dwtmm_stream() {
  return at::cuda::getCurrentCUDAStream().stream();
}

static inline int log2_if_pow2(int v) {
  return (v > 0 && (v & (v - 1)) == 0) ? __builtin_ctz((unsigned)v) : -1;
}

// C[M,N] = A[M,K] @ Bt[N,K]^T (+bias +relu), all bf16, fp32 accumulate.
void mfma_gemm(Tensor a, Tensor bt, Tensor bias, Tensor out, bool relu,
               bool has_bias) {
  const int64_t M = a.size(0);
  const int K = a.size(1);
  const int N = bt.size(0);
  TORCH_CHECK(bt.size(1) == K && out.size(0) == M && out.size(1) == N);
  dwtmm::ConvParams cp{};
  cp.M = M; cp.K = K; cp.Cout = N;
  const int mtiles = (M + dwtmm::BM - 1) / dwtmm::BM;
  const int ntiles = (N + dwtmm::BN - 1) / dwtmm::BN;
  const bool pipe = K >= 4 * dwtmm::BK;
  const bool swz = ntiles >= 8 && mtiles * ntiles >= 16;
  auto run3 = [&](auto reluc, auto biasc, auto pipec, auto swzc) {
    hipLaunchKernelGGL(
        (dwtmm::conv_implicit_gemm_kernel<dwtmm::A_DENSE,
                                          decltype(reluc)::value,
                                          decltype(biasc)::value,
                                          decltype(pipec)::value,
                                          decltype(swzc)::value>),
        dim3(mtiles * ntiles), dim3(dwtmm::THREADS), 0, dwtmm_stream(),
        (const c10::BFloat16*)a.data_ptr(), (const c10::BFloat16*)bt.data_ptr(),
        has_bias ? bias.data_ptr<float>() : nullptr,
        (c10::BFloat16*)out.data_ptr(), cp, mtiles, ntiles);
  };
  auto run = [&](auto reluc, auto biasc) {
    if (pipe) { if (swz) run3(reluc, biasc, std::true_type{}, std::true_type{});
                else run3(reluc, biasc, std::true_type{}, std::false_type{}); }
    else { if (swz) run3(reluc, biasc, std::false_type{}, std::true_type{});
           else run3(reluc, biasc, std::false_type{}, std::false_type{}); }
  };
  if (relu) { if (has_bias) run(std::true_type{}, std::true_type{});
              else run(std::true_type{}, std::false_type{}); }
  else { if (has_bias) run(std::false_type{}, std::true_type{});
         else run(std::false_type{}, std::false_type{}); }
}

// NHWC conv fwd: x (N,H,W,Cin) raw storage, wgt (Cout, KH*KW*Cin) raw
// channels_last storage, out (N,P,Q,Cout) raw storage.
void mfma_conv2d_fwd(Tensor x, Tensor wgt, Tensor bias, Tensor out,
                     int64_t N, int64_t H, int64_t W, int64_t Cin,
                     int64_t P, int64_t Q, int64_t Cout, int64_t KH,
                     int64_t KW, int64_t stride, int64_t pad, bool relu,
                     bool has_bias) {
  dwtmm::ConvParams cp{};
  cp.N = N; cp.H = H; cp.W = W; cp.Cin = Cin;
  cp.P = P; cp.Q = Q; cp.Cout = Cout;
  cp.KH = KH; cp.KW = KW; cp.stride = stride; cp.pad = pad;
  cp.M = N * P * Q;
  cp.K = KH * KW * Cin;
  cp.cShift = log2_if_pow2(Cin);
  const int mtiles = (cp.M + dwtmm::BM - 1) / dwtmm::BM;
  const int ntiles = (Cout + dwtmm::BN - 1) / dwtmm::BN;
  const bool gemm_fast = (KH == 1 && KW == 1 && stride == 1 && pad == 0 &&
                          Cin % 8 == 0);
  // pipelined staging pays only on the dense path (measured: the gather
  // path's extra register pressure under PIPE regressed l3.conv2 0.91->0.60)
  const bool pipe = gemm_fast && cp.K >= 4 * dwtmm::BK;
  const bool swz = gemm_fast && ntiles >= 8 && mtiles * ntiles >= 16;
  auto run = [&](auto fastc, auto reluc, auto biasc) {
    auto launch = [&](auto pipec, auto swzc) {
      hipLaunchKernelGGL(
          (dwtmm::conv_implicit_gemm_kernel<decltype(fastc)::value ? dwtmm::A_DENSE
                                                                   : dwtmm::A_CONV,
                                            decltype(reluc)::value,
                                            decltype(biasc)::value,
                                            decltype(pipec)::value,
                                            decltype(swzc)::value>),
          dim3(mtiles * ntiles), dim3(dwtmm::THREADS), 0, dwtmm_stream(),
          (const c10::BFloat16*)x.data_ptr(), (const c10::BFloat16*)wgt.data_ptr(),
          has_bias ? bias.data_ptr<float>() : nullptr,
          (c10::BFloat16*)out.data_ptr(), cp, mtiles, ntiles);
    };
    if (pipe) { if (swz) launch(std::true_type{}, std::true_type{});
                else launch(std::true_type{}, std::false_type{}); }
    else { if (swz) launch(std::false_type{}, std::true_type{});
           else launch(std::false_type{}, std::false_type{}); }
  };
  auto pick_rb = [&](auto fastc) {
    if (relu) { if (has_bias) run(fastc, std::true_type{}, std::true_type{});
                else run(fastc, std::true_type{}, std::false_type{}); }
    else { if (has_bias) run(fastc, std::false_type{}, std::true_type{});
           else run(fastc, std::false_type{}, std::false_type{}); }
  };
  if (gemm_fast) pick_rb(std::true_type{});
  else pick_rb(std::false_type{});
}


// dgrad: dx (N,H,W,Cin) = implicit-GEMM over dy (N,P,Q,Cdy) with
// wd (Cin, KH*KW*Cdy) = weight permuted to [ci][r][s][co] (host side).
void mfma_conv2d_dgrad(Tensor dy, Tensor wd, Tensor dx, int64_t N, int64_t H,
                       int64_t W, int64_t Cin, int64_t P, int64_t Q,
                       int64_t Cdy, int64_t KH, int64_t KW, int64_t stride,
                       int64_t pad) {
  dwtmm::ConvParams cp{};
  cp.N = N; cp.H = H; cp.W = W; cp.Cin = Cin;
  cp.P = P; cp.Q = Q; cp.Cout = Cin;  // GEMM output channels = Cin
  cp.KH = KH; cp.KW = KW; cp.stride = stride; cp.pad = pad;
  cp.Cdy = Cdy;
  cp.M = N * H * W;
  cp.K = KH * KW * Cdy;
  cp.cShift = log2_if_pow2(Cdy);
  const int mtiles = (cp.M + dwtmm::BM - 1) / dwtmm::BM;
  const int ntiles = (Cin + dwtmm::BN - 1) / dwtmm::BN;
  // 1x1/stride-1 dgrad IS the dense GEMM dy @ w^T — run it on the
  // pipelined/swizzled dense path instead of the per-chunk gather decode
  const bool gemm_fast = (KH == 1 && KW == 1 && stride == 1 && pad == 0 &&
                          Cdy % 8 == 0);
  const bool pipe = gemm_fast && cp.K >= 4 * dwtmm::BK;
  const bool swz = gemm_fast && ntiles >= 8 && mtiles * ntiles >= 16;
  auto launch = [&](auto modec, auto pipec, auto swzc) {
    hipLaunchKernelGGL(
        (dwtmm::conv_implicit_gemm_kernel<decltype(modec)::value,
                                          false, false,
                                          decltype(pipec)::value,
                                          decltype(swzc)::value>),
        dim3(mtiles * ntiles), dim3(dwtmm::THREADS), 0, dwtmm_stream(),
        (const c10::BFloat16*)dy.data_ptr(), (const c10::BFloat16*)wd.data_ptr(),
        nullptr, (c10::BFloat16*)dx.data_ptr(), cp, mtiles, ntiles);
  };
  using DEN = std::integral_constant<int, dwtmm::A_DENSE>;
  using DGR = std::integral_constant<int, dwtmm::A_DGRAD>;
  if (gemm_fast) {
    if (pipe) { if (swz) launch(DEN{}, std::true_type{}, std::true_type{});
                else launch(DEN{}, std::true_type{}, std::false_type{}); }
    else { if (swz) launch(DEN{}, std::false_type{}, std::true_type{});
           else launch(DEN{}, std::false_type{}, std::false_type{}); }
  } else if (stride > 1 && log2_if_pow2(Cdy) >= 0 && Cdy >= 8) {
    // class decomposition: one launch per (h%stride, w%stride) residue
    for (int ch = 0; ch < stride; ++ch) {
      for (int cw = 0; cw < stride; ++cw) {
        dwtmm::DgradClsParams gp{};
        gp.H = H; gp.W = W; gp.Cin = Cin; gp.P = P; gp.Q = Q;
        gp.Cdy = Cdy; gp.Kfull = KH * KW * Cdy;
        gp.ch = ch; gp.cw = cw; gp.stride = stride;
        gp.Hc = (int)((H - ch + stride - 1) / stride);
        gp.Wc = (int)((W - cw + stride - 1) / stride);
        gp.M = (int64_t)N * gp.Hc * gp.Wc;
        gp.cShift = log2_if_pow2(Cdy);
        int rl[16], sl[16], nr = 0, ns = 0;
        for (int r = 0; r < KH; ++r)
          if ((ch + pad - r) % stride == 0 && nr < 4) rl[nr++] = r;
        for (int s = 0; s < KW; ++s)
          if ((cw + pad - s) % stride == 0 && ns < 4) sl[ns++] = s;
        int ntaps = 0;
        for (int a = 0; a < nr && ntaps < 16; ++a)
          for (int b = 0; b < ns && ntaps < 16; ++b) {
            gp.dp[ntaps] = (int)((ch + pad - rl[a]) / stride);
            gp.dq[ntaps] = (int)((cw + pad - sl[b]) / stride);
            gp.rsIdx[ntaps] = rl[a] * KW + sl[b];
            ++ntaps;
          }
        gp.K = ntaps * Cdy;
        if (ntaps == 0 || gp.M == 0) continue;
        const int cm = (int)((gp.M + dwtmm::BM - 1) / dwtmm::BM);
        hipLaunchKernelGGL(dwtmm::dgrad_cls_kernel, dim3(cm * ntiles),
                           dim3(dwtmm::THREADS), 0, dwtmm_stream(),
                           (const c10::BFloat16*)dy.data_ptr(),
                           (const c10::BFloat16*)wd.data_ptr(),
                           (c10::BFloat16*)dx.data_ptr(), gp, cm, ntiles);
      }
    }
  } else {
    if (pipe) launch(DGR{}, std::true_type{}, std::false_type{});
    else launch(DGR{}, std::false_type{}, std::false_type{});
  }
}


// wgrad v2 (split-K): dy (N,P,Q,Cout) raw NHWC, x (N,H,W,Cin) raw NHWC,
// ws fp32 zero-initialized [Cout, KH*KW*Cin], dw bf16 channels_last weight
// storage (same [co][r][s][ci] flat layout as ws).
// ConvParams reuse here: cp.N = Cout (dy channel stride), cp.Cout =
// KH*KW*Cin (dw columns), cp.M = real NPQ.
void mfma_conv2d_wgrad(Tensor dy, Tensor x, Tensor dw, Tensor ws, int64_t N,
                       int64_t H, int64_t W, int64_t Cin, int64_t P,
                       int64_t Q, int64_t Cout, int64_t KH, int64_t KW,
                       int64_t stride, int64_t pad) {
  dwtmm::ConvParams cp{};
  cp.N = Cout; cp.H = H; cp.W = W; cp.Cin = Cin;
  cp.P = P; cp.Q = Q;
  cp.KH = KH; cp.KW = KW; cp.stride = stride; cp.pad = pad;
  cp.M = (int64_t)N * P * Q;
  cp.Cout = KH * KW * Cin;
  // 128 tile when both dims fill it (4x less cross-tile operand traffic);
  // 64 tile otherwise (stem/l1 shapes would waste half the MFMA work)
  const int WT = (Cout >= 128 && cp.Cout >= 128) ? 128 : 64;
  const int mtiles = (int)((Cout + WT - 1) / WT);
  const int ntiles = (cp.Cout + WT - 1) / WT;
  // split K so the grid lands at >=2048 workgroups (256 CUs, several waves
  // deep), k-slabs rounded to whole BK tiles
  int64_t splitk = std::max<int64_t>(1, 2048 / (mtiles * ntiles));
  const int64_t kt = (cp.M + dwtmm::WBK - 1) / dwtmm::WBK;  // total k-tiles
  splitk = std::min<int64_t>(splitk, kt);
  const int64_t klen = ((kt + splitk - 1) / splitk) * dwtmm::WBK;
  const bool aligned = (Cin % 64 == 0);
  auto launch = [&](auto ac, auto wtc) {
    hipLaunchKernelGGL(
        (dwtmm::wgrad_splitk_kernel<decltype(ac)::value, decltype(wtc)::value>),
        dim3(mtiles * ntiles * splitk), dim3(256), 0, dwtmm_stream(),
        (const c10::BFloat16*)dy.data_ptr(), (const c10::BFloat16*)x.data_ptr(),
        ws.data_ptr<float>(), cp, mtiles, ntiles, (int)splitk, klen);
  };
  using T64 = std::integral_constant<int, 64>;
  using T128 = std::integral_constant<int, 128>;
  if (aligned) { if (WT == 128) launch(std::true_type{}, T128{});
                 else launch(std::true_type{}, T64{}); }
  else { if (WT == 128) launch(std::false_type{}, T128{});
         else launch(std::false_type{}, T64{}); }
  const int64_t n = (int64_t)Cout * cp.Cout;
  hipLaunchKernelGGL(dwtmm::f32_to_bf16_kernel, dim3((n + 255) / 256),
                     dim3(256), 0, dwtmm_stream(), ws.data_ptr<float>(),
                     (c10::BFloat16*)dw.data_ptr(), n);
}
