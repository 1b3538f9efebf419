// Fused causal flash attention forward for CDNA4.
//
// Replaces the reference's six-op attention chain (GPTJ.py:164-191:
// matmul, where-mask, scale, softmax, dropout, matmul with a materialized
// fp32 [T,T] score tensor) with one MFMA kernel using online softmax
// (O(T) memory) and fp32 accumulation.
//
// v1 structure (correctness-first; the 8-wave swapped-QK^T structure of the
// CDNA4 guide is the optimization target for later passes):
//   grid  = (T/64, B*H); block = 256 threads = 4 waves
//   each wave owns 16 q rows; the block shares K/V tiles of 32 keys staged
//   in LDS; per kv-tile each wave computes S[16][32] with
//   v_mfma_f32_16x16x32_bf16 (contraction over D in 32-slices), does the
//   online-softmax update through wave-private LDS, and accumulates
//   O[16][D] with PV MFMAs.
//
// Fragment layouts validated on MI355X hardware (tools/mfma_probe.hip):
//   A[m][k]: lane l, slot j (0..7) -> A[l&15][(l>>4)*8 + j]
//   B[k][n]: lane l, slot j        -> B[(l>>4)*8 + j][l&15]
//   D[m][n]: lane l, reg r (0..3)  -> D[(l>>4)*4 + r][l&15]

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace samd {

typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

constexpr int QBLK = 64;   // q rows per block
constexpr int KBLK = 32;   // keys per kv tile
constexpr int WROWS = 16;  // q rows per wave

// Element strides of a [B, H, T, D] tensor with stride(D) == 1.  Kernels
// address through these, so transposed views ([B,T,H,D] physical — the
// model's natural projection layout) need NO .contiguous() copies, and the
// output is written straight into [B,T,H,D] physical so the model's
// transpose+reshape after attention is a free view.
struct TStr {
  long sb, sh, st;
};
static TStr str_of(const at::Tensor& t) {
  return TStr{t.stride(0), t.stride(1), t.stride(2)};
}

// LDS layout (per block), bf16 K/V tiles + per-wave P relayout buffer:
//   K[32][D], V[32][D] (+8 bf16 row pad: row stride = 16 B mod 256 B, so
//   the 16-lane b128 groups of the K-fragment reads land on distinct slots)
//   per wave: P[16][32] bf16 (score D-layout -> A-fragment relayout).
// Softmax state (m, l) lives in registers: row r of a wave's 16 q rows is
// owned by quadrant r>>2 at accumulator reg r&3; its 16 score columns live
// in that quadrant's 16 lanes, so row max/sum reduce with 4 shfl_xor steps
// and never touch LDS.

// ---------------------------------------------------------------------------
// ds_read_b64_tr_b16 B-fragment read (hardware transpose).
// Semantics pinned by tools/tr_probe.hip on MI355X: with per-lane source
// address a(s), lane l's element j comes from element (l&3) of the aligned
// 4-element window of lane s = (l&~15) + 4*j + ((l>>2)&3) in its 16-lane
// group.  For the MFMA B-fragment B[(l>>4)*8+j][dcol=l&15] of a row-major
// LDS tile, each lane addresses row (g*8 + ((l>>2)&3)), col
// (col0 + 4*(l&3)) — two reads (rows +0, +4) yield the 8-slot fragment
// with no bank-conflict-prone scalar gathers.  Addresses must be 8-B
// aligned (row strides here are multiples of 8 B — G17).
// ---------------------------------------------------------------------------
typedef __attribute__((ext_vector_type(4))) short bf16x4;

__device__ __forceinline__ unsigned lds_addr32(const unsigned short* p) {
  return (unsigned)(unsigned long long)(
      __attribute__((address_space(3))) const unsigned short*)p;
}

template <int P1, int P2>
__device__ __forceinline__ void tr16_bfrag2(const unsigned short* t1_row0,
                                            const unsigned short* t2_row0,
                                            int lane, bf16x8& o1, bf16x8& o2) {
  const int src_row = (lane >> 2) & 3;
  const int src_col = 4 * (lane & 3);
  const unsigned a0 = lds_addr32(t1_row0 + src_row * P1 + src_col);
  const unsigned a1 = a0 + 4u * P1 * 2u;
  const unsigned b0 = lds_addr32(t2_row0 + src_row * P2 + src_col);
  const unsigned b1 = b0 + 4u * P2 * 2u;
  bf16x4 v0, v1, w0, w1;
  asm volatile(
      "ds_read_b64_tr_b16 %0, %4\n\t"
      "ds_read_b64_tr_b16 %1, %5\n\t"
      "ds_read_b64_tr_b16 %2, %6\n\t"
      "ds_read_b64_tr_b16 %3, %7\n\t"
      "s_waitcnt lgkmcnt(0)"
      : "=&v"(v0), "=&v"(v1), "=&v"(w0), "=&v"(w1)
      : "v"(a0), "v"(a1), "v"(b0), "v"(b1)
      : "memory");
  o1 = __builtin_shufflevector(v0, v1, 0, 1, 2, 3, 4, 5, 6, 7);
  o2 = __builtin_shufflevector(w0, w1, 0, 1, 2, 3, 4, 5, 6, 7);
}

template <int ROWPITCH>
__device__ __forceinline__ bf16x8 tr16_bfrag(const unsigned short* tile_row0,
                                             int lane) {
  // tile_row0: &tile[base_row][col0] for this fragment's 8-row block.
  const int src_row = (lane >> 2) & 3;
  const int src_col = 4 * (lane & 3);
  const unsigned a0 =
      lds_addr32(tile_row0 + src_row * ROWPITCH + src_col);
  const unsigned a1 = a0 + 4u * ROWPITCH * 2u;
  bf16x4 v0, v1;
  asm volatile(
      "ds_read_b64_tr_b16 %0, %2\n\t"
      "ds_read_b64_tr_b16 %1, %3\n\t"
      "s_waitcnt lgkmcnt(0)"
      : "=&v"(v0), "=&v"(v1)
      : "v"(a0), "v"(a1)
      : "memory");
  return __builtin_shufflevector(v0, v1, 0, 1, 2, 3, 4, 5, 6, 7);
}

template <int D>
struct AttnLds {
  unsigned short k[KBLK][D + 8];
  unsigned short v[KBLK][D + 8];
  unsigned short p[4][WROWS][KBLK];
};

template <int D>
__launch_bounds__(256, 2)
__global__ void attn_fwd_kernel(const unsigned short* __restrict__ Q,
                                const unsigned short* __restrict__ K,
                                const unsigned short* __restrict__ V,
                                unsigned short* __restrict__ O,
                                float* __restrict__ LSE, int T, int n_heads,
                                int n_kv, float scale, int causal,
                                int kv_len, TStr qs, TStr ks, TStr vs,
                                TStr os) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  AttnLds<D>& lds = *reinterpret_cast<AttnLds<D>*>(smem);

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  // grid = (B*H, T/QBLK): bh varies fastest so the causal q-block work
  // imbalance spreads across XCDs (dispatcher places block b on XCD b%8;
  // with q-block fastest, XCD k would get ONLY q-block-index k's length)
  const int q0_block = blockIdx.y * QBLK;        // block's first q row
  const int q0 = q0_block + wid * WROWS;         // wave's first q row
  const long bh = blockIdx.x;
  const long b = bh / n_heads, h = bh % n_heads;
  const long h_kv = h * n_kv / n_heads;  // GQA: q heads share kv heads
  const unsigned short* Qh = Q + b * qs.sb + h * qs.sh;
  const unsigned short* Kh = K + b * ks.sb + h_kv * ks.sh;
  const unsigned short* Vh = V + b * vs.sb + h_kv * vs.sh;

  const int r = lane & 15;       // fragment row/col index
  const int qg = lane >> 4;      // quadrant 0..3

  // ---- load Q fragments to registers: frag_q[ds][j] = Q[q0 + r][ds*32 + qg*8 + j]
  bf16x8 frag_q[D / 32];
#pragma unroll
  for (int ds = 0; ds < D / 32; ++ds) {
    const unsigned short* src = Qh + (long)(q0 + r) * qs.st + ds * 32 + qg * 8;
    frag_q[ds] = *reinterpret_cast<const bf16x8*>(src);
  }

  // ---- O accumulators: otile[dt][reg] = O[(qg*4+reg)][dt*16 + r]
  f32x4 otile[D / 16];
#pragma unroll
  for (int dt = 0; dt < D / 16; ++dt) otile[dt] = {0.f, 0.f, 0.f, 0.f};

  // ---- softmax state in registers (per quadrant: rows qg*4+reg)
  float m_reg[4] = {-INFINITY, -INFINITY, -INFINITY, -INFINITY};
  float l_reg[4] = {0.f, 0.f, 0.f, 0.f};

  const int kv_end = causal ? (q0_block + QBLK) : T;
  // ---- async-stage split (guide T14): each thread holds the NEXT tile's
  // K/V rows in registers; the global loads for tile i+1 are issued before
  // tile i's compute so HBM latency hides under the MFMAs, and the LDS
  // write happens right after the tile-consumed barrier.
  constexpr int CHUNKS = (KBLK * D) / (256 * 8);
  bf16x8 stg_k[CHUNKS], stg_v[CHUNKS];
#pragma unroll
  for (int c = 0; c < CHUNKS; ++c) {
    const int idx = (c * 256 + threadIdx.x) * 8;
    const int row = idx / D, col = idx % D;
    stg_k[c] = *reinterpret_cast<const bf16x8*>(Kh + (long)row * ks.st + col);
    stg_v[c] = *reinterpret_cast<const bf16x8*>(Vh + (long)row * vs.st + col);
  }
  for (int kv0 = 0; kv0 < kv_end; kv0 += KBLK) {
    __syncthreads();  // previous tile fully consumed
#pragma unroll
    for (int c = 0; c < CHUNKS; ++c) {
      const int idx = (c * 256 + threadIdx.x) * 8;
      const int row = idx / D, col = idx % D;
      *reinterpret_cast<bf16x8*>(&lds.k[row][col]) = stg_k[c];
      *reinterpret_cast<bf16x8*>(&lds.v[row][col]) = stg_v[c];
    }
    if (kv0 + KBLK < kv_end) {
#pragma unroll
      for (int c = 0; c < CHUNKS; ++c) {
        const int idx = (c * 256 + threadIdx.x) * 8;
        const int row = idx / D, col = idx % D;
        stg_k[c] = *reinterpret_cast<const bf16x8*>(
            Kh + (long)(kv0 + KBLK + row) * ks.st + col);
        stg_v[c] = *reinterpret_cast<const bf16x8*>(
            Vh + (long)(kv0 + KBLK + row) * vs.st + col);
      }
    }
    __syncthreads();  // staged tile visible

    // ---- S[16][32] = scale * Q K^T  (two 16-col MFMA tiles)
    f32x4 acc0 = {0.f, 0.f, 0.f, 0.f};
    f32x4 acc1 = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int ds = 0; ds < D / 32; ++ds) {
      bf16x8 bk0 = *reinterpret_cast<const bf16x8*>(&lds.k[r][ds * 32 + qg * 8]);
      bf16x8 bk1 =
          *reinterpret_cast<const bf16x8*>(&lds.k[16 + r][ds * 32 + qg * 8]);
      acc0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(frag_q[ds], bk0, acc0, 0, 0, 0);
      acc1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(frag_q[ds], bk1, acc1, 0, 0, 0);
    }
    // ---- online softmax fully in registers.  Row (qg*4+reg)'s 32
    // scores sit in acc0[reg]/acc1[reg] across the quadrant's 16 lanes;
    // reduce with 4 shfl_xor steps (no LDS round trip).
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int row = qg * 4 + reg;
      const int qrow = q0 + row;
      float s0 = acc0[reg] * scale;
      float s1 = acc1[reg] * scale;
      if (causal) {
        if (kv0 + r > qrow) s0 = -INFINITY;
        if (kv0 + 16 + r > qrow) s1 = -INFINITY;
      }
      // ragged T: keys beyond the unpadded length contribute nothing
      if (kv0 + r >= kv_len) s0 = -INFINITY;
      if (kv0 + 16 + r >= kv_len) s1 = -INFINITY;
      float tm = fmaxf(s0, s1);
#pragma unroll
      for (int off = 1; off < 16; off <<= 1)
        tm = fmaxf(tm, __shfl_xor(tm, off, 64));
      const float m_new = fmaxf(m_reg[reg], tm);
      const float al =
          (m_reg[reg] == -INFINITY) ? 0.f : __expf(m_reg[reg] - m_new);
      const float p0 = (s0 == -INFINITY) ? 0.f : __expf(s0 - m_new);
      const float p1 = (s1 == -INFINITY) ? 0.f : __expf(s1 - m_new);
      float ps = p0 + p1;
#pragma unroll
      for (int off = 1; off < 16; off <<= 1)
        ps += __shfl_xor(ps, off, 64);
      l_reg[reg] = l_reg[reg] * al + ps;
      m_reg[reg] = m_new;
      lds.p[wid][row][r] = f2us(p0);
      lds.p[wid][row][16 + r] = f2us(p1);
#pragma unroll
      for (int dt = 0; dt < D / 16; ++dt) otile[dt][reg] *= al;
    }

    // ---- PV: O[16][D] += P[16][32] x V[32][D]
    bf16x8 frag_p = *reinterpret_cast<const bf16x8*>(&lds.p[wid][r][qg * 8]);
#pragma unroll
    for (int dt = 0; dt < D / 16; ++dt) {
      bf16x8 bv = tr16_bfrag<D + 8>(&lds.v[qg * 8][dt * 16], lane);
      otile[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(frag_p, bv, otile[dt], 0, 0, 0);
    }
  }

  // ---- epilogue: O /= l, write bf16; LSE = m + log(l)
  unsigned short* Oh = O + b * os.sb + h * os.sh;
#pragma unroll
  for (int reg = 0; reg < 4; ++reg) {
    int row = qg * 4 + reg;
    float linv = 1.f / l_reg[reg];
#pragma unroll
    for (int dt = 0; dt < D / 16; ++dt)
      Oh[(long)(q0 + row) * os.st + dt * 16 + r] =
          f2us(otile[dt][reg] * linv);
  }
  if (r == 0 && LSE != nullptr) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg)
      LSE[bh * (long)T + q0 + qg * 4 + reg] =
          m_reg[reg] + __logf(l_reg[reg]);
  }
}

// ===========================================================================
// Forward v2 (opt-in: SAMD_ATTN_V2=1): the CDNA4-guide structure — swapped
// QK^T on v_mfma_f32_32x32x16_bf16 so each lane's 16 score values belong to
// ONE q row (lane-local online softmax, one shfl_xor(32) merge), P stays in
// registers and reaches the PV A-fragment via permlane32_swap half
// exchanges; V rides the tr16 transpose read.  2x the q rows per MFMA of
// the v1 16x16 structure.  Layouts validated by tools/mfma_probe.hip:
//   32x32x16  A[m][k]: lane l, j -> A[l&31][(l>>5)*8+j]
//             B[k][n]: lane l, j -> B[(l>>5)*8+j][l&31]
//             D[m][n]: lane l, reg -> D[(reg&3)+8*(reg>>2)+4*(l>>5)][l&31]
// ===========================================================================


typedef __attribute__((ext_vector_type(16))) float f32x16;

template <int D>
struct AttnV2Lds {
  unsigned short k[KBLK][D + 8];
  unsigned short v[KBLK][D + 8];
};

template <int D>
__launch_bounds__(256, 2)
__global__ void attn_fwd_v2_kernel(const unsigned short* __restrict__ Q,
                                   const unsigned short* __restrict__ K,
                                   const unsigned short* __restrict__ V,
                                   unsigned short* __restrict__ O,
                                   float* __restrict__ LSE, int T,
                                   int n_heads, int n_kv, float scale,
                                   int causal, int kv_len, TStr qs,
                                   TStr ks, TStr vs,
                                   TStr os) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  AttnV2Lds<D>& lds = *reinterpret_cast<AttnV2Lds<D>*>(smem);

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int qr = lane & 31;       // this lane's q row (within the wave's 32)
  const int hi = lane >> 5;       // half index
  const long bh = blockIdx.x;
  const long b = bh / n_heads, h = bh % n_heads;
  const long h_kv = h * n_kv / n_heads;  // GQA
  const int q0_block = blockIdx.y * (4 * 32);
  const int q0 = q0_block + wid * 32;
  const unsigned short* Qh = Q + b * qs.sb + h * qs.sh;
  const unsigned short* Kh = K + b * ks.sb + h_kv * ks.sh;
  const unsigned short* Vh = V + b * vs.sb + h_kv * vs.sh;

  // Q fragments (B operand of swapped QK^T): frag_q[ds][j] =
  // Q[q0+qr][ds*16 + hi*8 + j].  At D=256 the preloaded array would push
  // the kernel to 256 VGPRs with spills; the QK loop re-reads Q from L2
  // per tile instead (the wave's 16 KB stays hot).
  constexpr bool QREG = (D < 256);
  bf16x8 frag_q[QREG ? D / 16 : 1];
  if constexpr (QREG) {
#pragma unroll
    for (int ds = 0; ds < D / 16; ++ds)
      frag_q[ds] = *reinterpret_cast<const bf16x8*>(
          Qh + (long)(q0 + qr) * qs.st + ds * 16 + hi * 8);
  }

  // O accumulators: oacc[dv][reg] = O[(reg&3)+8*(reg>>2)+4*hi][dv*32 + qr]
  f32x16 oacc[D / 32];
#pragma unroll
  for (int dv = 0; dv < D / 32; ++dv)
#pragma unroll
    for (int e = 0; e < 16; ++e) oacc[dv][e] = 0.f;

  float m_row = -INFINITY, l_row = 0.f;

  // staging registers (T14 split)
  constexpr int CHUNKS = (KBLK * D) / (256 * 8);
  bf16x8 stg_k[CHUNKS], stg_v[CHUNKS];
#pragma unroll
  for (int c = 0; c < CHUNKS; ++c) {
    const int idx = (c * 256 + threadIdx.x) * 8;
    const int row = idx / D, col = idx % D;
    stg_k[c] = *reinterpret_cast<const bf16x8*>(Kh + (long)row * ks.st + col);
    stg_v[c] = *reinterpret_cast<const bf16x8*>(Vh + (long)row * vs.st + col);
  }

  const int kv_end = causal ? (q0_block + 4 * 32) : T;
  for (int kv0 = 0; kv0 < kv_end; kv0 += KBLK) {
    __syncthreads();
#pragma unroll
    for (int c = 0; c < CHUNKS; ++c) {
      const int idx = (c * 256 + threadIdx.x) * 8;
      const int row = idx / D, col = idx % D;
      *reinterpret_cast<bf16x8*>(&lds.k[row][col]) = stg_k[c];
      *reinterpret_cast<bf16x8*>(&lds.v[row][col]) = stg_v[c];
    }
    if (kv0 + KBLK < kv_end) {
#pragma unroll
      for (int c = 0; c < CHUNKS; ++c) {
        const int idx = (c * 256 + threadIdx.x) * 8;
        const int row = idx / D, col = idx % D;
        stg_k[c] = *reinterpret_cast<const bf16x8*>(
            Kh + (long)(kv0 + KBLK + row) * ks.st + col);
        stg_v[c] = *reinterpret_cast<const bf16x8*>(
            Vh + (long)(kv0 + KBLK + row) * vs.st + col);
      }
    }
    __syncthreads();

    // ---- S^T[key][q] = K Q^T: one 32x32 tile over the D contraction
    f32x16 st;
#pragma unroll
    for (int e = 0; e < 16; ++e) st[e] = 0.f;
#pragma unroll
    for (int ds = 0; ds < D / 16; ++ds) {
      bf16x8 ak = *reinterpret_cast<const bf16x8*>(
          &lds.k[qr][ds * 16 + hi * 8]);
      bf16x8 fq;
      if constexpr (QREG)
        fq = frag_q[ds];
      else
        fq = *reinterpret_cast<const bf16x8*>(
            Qh + (long)(q0 + qr) * qs.st + ds * 16 + hi * 8);
      st = __builtin_amdgcn_mfma_f32_32x32x16_bf16(ak, fq, st, 0, 0, 0);
    }

    // ---- lane-local online softmax for q row (q0 + qr).
    // lane holds keys key16(reg) = (reg&3) + 8*(reg>>2) + 4*hi.
    const int q_glob = q0 + qr;
    float p[16];
    float tile_max = -INFINITY;
#pragma unroll
    for (int reg = 0; reg < 16; ++reg) {
      const int key_loc = (reg & 3) + 8 * (reg >> 2) + 4 * hi;
      float sv = st[reg] * scale;
      if ((causal && kv0 + key_loc > q_glob) || kv0 + key_loc >= kv_len)
        sv = -INFINITY;
      p[reg] = sv;
      tile_max = fmaxf(tile_max, sv);
    }
    tile_max = fmaxf(tile_max, __shfl_xor(tile_max, 32, 64));
    const float m_new = fmaxf(m_row, tile_max);
    const float alpha = (m_row == -INFINITY) ? 0.f : __expf(m_row - m_new);
    float psum = 0.f;
#pragma unroll
    for (int reg = 0; reg < 16; ++reg) {
      const float pv = (p[reg] == -INFINITY) ? 0.f : __expf(p[reg] - m_new);
      p[reg] = pv;
      psum += pv;
    }
    psum += __shfl_xor(psum, 32, 64);
    l_row = l_row * alpha + psum;
    m_row = m_new;

    // ---- P (f32, keys in D-layout order) -> PV A-fragments via pack +
    // permlane32_swap.  pk[i] packs keys (2i, 2i+1) of this lane's set:
    //   lo half (hi=0): keys {0,1},{2,3},{8,9},{10,11},{16,17},...
    //   hi half (hi=1): keys {4,5},{6,7},{12,13},{14,15},...
    // after swap(pk[2g], pk[2g+1]? see below) each lane holds the pairs of
    // ITS A-fragment rows: frag keys (l>>5)*8 + 0..7 per 16-key tile.
    unsigned pk[8];
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      const unsigned lo = f2us(p[2 * i]);
      const unsigned hi2 = f2us(p[2 * i + 1]);
      pk[i] = lo | (hi2 << 16);
    }
    // tile g (g=0: keys 0-15, g=1: keys 16-31): lane pairs pk[4g+0..3] hold
    //   hi=0: (0,1),(2,3),(8,9),(10,11)   [+16 for g=1]
    //   hi=1: (4,5),(6,7),(12,13),(14,15)
    // swap(pk[4g+0], pk[4g+2]): lo lane gets {(0,1) | (4,5)->? }
    bf16x8 pa[2];
#pragma unroll
    for (int g = 0; g < 2; ++g) {
      auto r0 = __builtin_amdgcn_permlane32_swap(pk[4 * g + 0], pk[4 * g + 2],
                                                 false, false);
      auto r1 = __builtin_amdgcn_permlane32_swap(pk[4 * g + 1], pk[4 * g + 3],
                                                 false, false);
      // r0[0]: lo half keeps (0,1), hi half got lo's (8,9)
      // r0[1]: lo half got hi's (4,5), hi half keeps (12,13)
      // r1[0]: (2,3) / lo's (10,11);  r1[1]: hi's (6,7) / (14,15)
      // A-frag for this lane: keys hi*8 + {0..7} =
      //   lo: (0,1),(2,3),(4,5),(6,7) = r0[0], r1[0], r0[1], r1[1]
      //   hi: (8,9),(10,11),(12,13),(14,15) = r0[0], r1[0], r0[1], r1[1]
      unsigned w0 = r0[0], w1 = r1[0], w2 = r0[1], w3 = r1[1];
      pa[g][0] = (short)(w0 & 0xffff);
      pa[g][1] = (short)(w0 >> 16);
      pa[g][2] = (short)(w1 & 0xffff);
      pa[g][3] = (short)(w1 >> 16);
      pa[g][4] = (short)(w2 & 0xffff);
      pa[g][5] = (short)(w2 >> 16);
      pa[g][6] = (short)(w3 & 0xffff);
      pa[g][7] = (short)(w3 >> 16);
    }

    // ---- rescale O by alpha of each accumulated row (rows differ from
    // this lane's softmax row): alpha[row] sits in lanes row and row+32.
#pragma unroll
    for (int reg = 0; reg < 16; ++reg) {
      const int row = (reg & 3) + 8 * (reg >> 2) + 4 * hi;
      const float al = __shfl(alpha, row, 64);
#pragma unroll
      for (int dv = 0; dv < D / 32; ++dv) oacc[dv][reg] *= al;
    }

    // ---- PV: O^T? no — O[q][d] += P[q][k] V[k][d]:
    // mfma(A=pa (32 q x 16 k), B=V[k][d] 16x32) per (key half g, dv)
#pragma unroll
    for (int dv = 0; dv < D / 32; ++dv) {
#pragma unroll
      for (int g = 0; g < 2; ++g) {
        // B fragment via tr16: B[kk=hi*8+j][dcol=l&31]; group-dependent
        // base col handled through the per-lane source address
        const int src_row = g * 16 + ((lane >> 4) & 2) * 4 + ((lane >> 2) & 3);
        const int src_col = dv * 32 + ((lane >> 4) & 1) * 16 + 4 * (lane & 3);
        const unsigned a0 = lds_addr32(&lds.v[src_row][src_col]);
        const unsigned a1 = a0 + 4u * (D + 8) * 2u;
        bf16x4 v0, v1;
        asm volatile(
            "ds_read_b64_tr_b16 %0, %2\n\t"
            "ds_read_b64_tr_b16 %1, %3\n\t"
            "s_waitcnt lgkmcnt(0)"
            : "=&v"(v0), "=&v"(v1)
            : "v"(a0), "v"(a1)
            : "memory");
        bf16x8 bv = __builtin_shufflevector(v0, v1, 0, 1, 2, 3, 4, 5, 6, 7);
        oacc[dv] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa[g], bv,
                                                           oacc[dv], 0, 0, 0);
      }
    }
  }

  // ---- epilogue
  unsigned short* Oh = O + b * os.sb + h * os.sh;
  const float linv_own = 1.f / l_row;
#pragma unroll
  for (int reg = 0; reg < 16; ++reg) {
    const int row = (reg & 3) + 8 * (reg >> 2) + 4 * hi;
    const float linv = __shfl(linv_own, row, 64);
#pragma unroll
    for (int dv = 0; dv < D / 32; ++dv)
      Oh[(long)(q0 + row) * os.st + dv * 32 + qr] = f2us(oacc[dv][reg] * linv);
  }
  if (hi == 0 && LSE != nullptr)
    LSE[bh * (long)T + q0 + qr] = m_row + __logf(l_row);
}



// ---------------------------------------------------------------------------
// Host wrapper
// ---------------------------------------------------------------------------
std::vector<at::Tensor> attn_fwd(at::Tensor q, at::Tensor k, at::Tensor v,
                                 bool causal, int64_t kv_len) {
  // kv_len < T marks ragged sequences: q/k/v are zero-padded to a 64
  // multiple by the wrapper and keys >= kv_len are masked out in-kernel
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == at::kBFloat16,
              "attn_fwd: bf16 only");
  TORCH_CHECK(q.dim() == 4, "attn_fwd: [B, H, T, D]");
  TORCH_CHECK(q.stride(3) == 1 && k.stride(3) == 1 && v.stride(3) == 1,
              "attn_fwd: innermost dim must be dense");
  const int B = (int)q.size(0), H = (int)q.size(1), T = (int)q.size(2),
            D = (int)q.size(3);
  const int Hkv = (int)k.size(1);
  TORCH_CHECK(H % Hkv == 0, "attn_fwd: q heads must be a multiple of kv heads");
  TORCH_CHECK(k.size(2) == T, "attn_fwd: q/k length mismatch");
  TORCH_CHECK(T % QBLK == 0, "attn_fwd: T must be a multiple of 64");
  if (kv_len <= 0) kv_len = T;
  TORCH_CHECK(kv_len <= T, "attn_fwd: kv_len exceeds padded T");
  TORCH_CHECK(D == 64 || D == 128 || D == 256, "attn_fwd: D in {64,128,256}");
  TORCH_CHECK((reinterpret_cast<uintptr_t>(q.data_ptr()) & 15) == 0 &&
              q.stride(2) % 8 == 0 && k.stride(2) % 8 == 0 &&
              v.stride(2) % 8 == 0, "attn_fwd: rows must be 16-B aligned");

  // output physically [B, T, H, D]: the model's post-attention
  // transpose(1,2).reshape is then a zero-copy view
  auto o_phys = at::empty({B, T, H, D}, q.options());
  auto o = o_phys.permute({0, 2, 1, 3});
  auto lse = at::empty({B, H, T}, q.options().dtype(at::kFloat));
  auto stream = at::hip::getCurrentHIPStream();
  const float scale = 1.f / sqrtf((float)D);

  // v2 structure (swapped 32x32 MFMAs, in-register P): default ON for the
  // shapes it supports (+40% measured at D=128/64); SAMD_ATTN_V2=0 forces
  // the v1 16x16 structure.
  static const bool use_v2 = [] {
    const char* e = getenv("SAMD_ATTN_V2");
    return !(e && e[0] == '0');
  }();
  // D=256 would spill at this structure (256 VGPRs + 53); v1 handles it
  if (use_v2 && T % 128 == 0 && D <= 128) {
    dim3 grid2(B * H, T / 128), block2(256);
#define LAUNCH_V2(DD)                                                        \
    do {                                                                     \
      size_t shmem = sizeof(AttnV2Lds<DD>);                                  \
      hipLaunchKernelGGL((attn_fwd_v2_kernel<DD>), grid2, block2, shmem,     \
                         stream.stream(),                                    \
                         reinterpret_cast<const unsigned short*>(q.data_ptr()), \
                         reinterpret_cast<const unsigned short*>(k.data_ptr()), \
                         reinterpret_cast<const unsigned short*>(v.data_ptr()), \
                         reinterpret_cast<unsigned short*>(o.data_ptr()),    \
                         lse.data_ptr<float>(), T, H, Hkv, scale,            \
                         causal ? 1 : 0, (int)kv_len, str_of(q), str_of(k),  \
                         str_of(v), str_of(o));                              \
    } while (0)
    if (D == 64) LAUNCH_V2(64);
    else LAUNCH_V2(128);
#undef LAUNCH_V2
    return {o, lse};
  }
  dim3 grid(B * H, T / QBLK), block(256);

#define LAUNCH(DD)                                                          \
  do {                                                                      \
    size_t shmem = sizeof(AttnLds<DD>);                                     \
    hipLaunchKernelGGL((attn_fwd_kernel<DD>), grid, block, shmem,           \
                       stream.stream(),                                     \
                       reinterpret_cast<const unsigned short*>(q.data_ptr()), \
                       reinterpret_cast<const unsigned short*>(k.data_ptr()), \
                       reinterpret_cast<const unsigned short*>(v.data_ptr()), \
                       reinterpret_cast<unsigned short*>(o.data_ptr()),     \
                       lse.data_ptr<float>(), T, H, Hkv, scale,             \
                       causal ? 1 : 0, (int)kv_len, str_of(q), str_of(k),   \
                       str_of(v), str_of(o));                               \
  } while (0)

  if (D == 64) LAUNCH(64);
  else if (D == 128) LAUNCH(128);
  else LAUNCH(256);
#undef LAUNCH
  return {o, lse};
}

}  // namespace samd

// ===========================================================================
// Backward (FA2 scheme): grid over kv tiles; each block owns 64 keys
// (16 per wave) and loops over q tiles of 32 rows, recomputing P from the
// saved LSE.  dK/dV accumulate in registers across the q loop; dQ partials
// go to a fp32 buffer with atomics (summed across kv blocks).
//   dV = P^T dO;  dP^T = V dO^T;  dS^T = P^T o (dP^T - delta) * scale;
//   dK = dS^T Q;  dQ = dS K;  delta = rowsum(dO o O) (computed by caller).
// ===========================================================================

namespace samd {

namespace bwd {
constexpr int KB = 64;   // keys per block
constexpr int QT = 32;   // q rows per tile
}

// V placement is D-dependent: at D=256 the full struct would be ~112 KB
// (one block/CU), so V stays in L2 and its wave-private A-fragments read
// global 16-B vectors — measured +39% at the GPT-J backward shape.  At
// D<=128 the struct fits two blocks/CU WITH the V tile, and the LDS copy
// is faster (keeping it avoided a measured regression at T=2048/D=128).
template <int D, bool HASV>
struct AttnBwdLds {
  unsigned short kt[bwd::KB][D + 8];
  unsigned short vt[HASV ? bwd::KB : 1][D + 8];
  unsigned short qt[bwd::QT][D + 8];
  unsigned short dot[bwd::QT][D + 8];
  unsigned short pt[4][16][bwd::QT + 8];   // per-wave P^T tile
  unsigned short dst[bwd::KB][bwd::QT + 8]; // shared dS^T (bf16, scaled)
  float lse_t[bwd::QT];
  float delta_t[bwd::QT];
};


// delta = rowsum(dO * O) for the FA2 backward, one fused pass.  The torch
// composition (dout.float() * o.float()).sum(-1) materializes two fp32
// copies and a product tensor (~7x the minimal bytes); at 28 layers that
// wrapper cost ~6% of the whole training step.  One wave per (b,h,t) row,
// vectorized bf16x8 loads, fp32 accumulate.
template <int D>
__global__ void attn_delta_kernel(const unsigned short* __restrict__ dO,
                                  const unsigned short* __restrict__ O,
                                  float* __restrict__ delta, int T,
                                  int n_heads, TStr dos, TStr os) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const long bh = blockIdx.y;
  const long b = bh / n_heads, h = bh % n_heads;
  const unsigned short* dOh = dO + b * dos.sb + h * dos.sh;
  const unsigned short* Oh = O + b * os.sb + h * os.sh;
  constexpr int EPL = D / 64;  // elements per lane (>= 1)
  for (int t = blockIdx.x * 4 + wid; t < T; t += gridDim.x * 4) {
    float acc = 0.f;
    if constexpr (EPL >= 4) {
#pragma unroll
      for (int c = 0; c < EPL / 4; ++c) {
        const int off = (lane * (EPL / 4) + c * 64) * 4;
        bf16x4 a = *reinterpret_cast<const bf16x4*>(dOh + (long)t * dos.st + off);
        bf16x4 o4 = *reinterpret_cast<const bf16x4*>(Oh + (long)t * os.st + off);
#pragma unroll
        for (int e = 0; e < 4; ++e)
          acc += us2f((unsigned short)a[e]) * us2f((unsigned short)o4[e]);
      }
    } else {
#pragma unroll
      for (int e = 0; e < EPL; ++e) {
        const int off = lane * EPL + e;
        acc += us2f(dOh[(long)t * dos.st + off]) *
               us2f(Oh[(long)t * os.st + off]);
      }
    }
    acc = wave_allsum(acc);
    if (lane == 0) delta[bh * (long)T + t] = acc;
  }
}

template <int D, bool TR16, bool VL2>
__launch_bounds__(256, 2)
__global__ void attn_bwd_kernel(const unsigned short* __restrict__ dO,
                                const unsigned short* __restrict__ Q,
                                const unsigned short* __restrict__ K,
                                const unsigned short* __restrict__ V,
                                const float* __restrict__ LSE,
                                const float* __restrict__ DELTA,
                                unsigned short* __restrict__ dK,
                                unsigned short* __restrict__ dV,
                                float* __restrict__ dKf,
                                float* __restrict__ dVf, int T, int n_heads,
                                int n_kv, float scale, int causal,
                                int kv_len, TStr dos,
                                TStr qs, TStr ks, TStr vs,
                                TStr dks, TStr dvs) {
  using namespace bwd;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  constexpr bool HASV = (D <= 128) && !VL2;  // V tile LDS-resident?
  AttnBwdLds<D, HASV>& lds = *reinterpret_cast<AttnBwdLds<D, HASV>*>(smem);

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int r = lane & 15;
  const int qg = lane >> 4;
  // bh fastest (see forward): spreads the causal kv-block imbalance
  const int kv0 = blockIdx.y * KB;
  const long bh = blockIdx.x;
  const long b = bh / n_heads, h = bh % n_heads;
  const long h_kv = h * n_kv / n_heads;  // GQA: q heads share kv heads
  const unsigned short* Qh = Q + b * qs.sb + h * qs.sh;
  const unsigned short* Kh = K + b * ks.sb + h_kv * ks.sh;
  const unsigned short* Vh = V + b * vs.sb + h_kv * vs.sh;
  const unsigned short* dOh = dO + b * dos.sb + h * dos.sh;
  const float* lse_h = LSE + bh * (long)T;
  const float* del_h = DELTA + bh * (long)T;

  // ---- stage the K tile once (and V when it fits; see struct comment)
  {
    constexpr int CHUNKS = (KB * D) / (256 * 8);
#pragma unroll
    for (int c = 0; c < CHUNKS; ++c) {
      int idx = (c * 256 + threadIdx.x) * 8;
      int row = idx / D, col = idx % D;
      *reinterpret_cast<bf16x8*>(&lds.kt[row][col]) =
          *reinterpret_cast<const bf16x8*>(
              Kh + (long)(kv0 + row) * ks.st + col);
      if constexpr (HASV)
        *reinterpret_cast<bf16x8*>(&lds.vt[row][col]) =
            *reinterpret_cast<const bf16x8*>(
                Vh + (long)(kv0 + row) * vs.st + col);
    }
  }

  f32x4 acc_dv[D / 16], acc_dk[D / 16];
#pragma unroll
  for (int dt = 0; dt < D / 16; ++dt) {
    acc_dv[dt] = {0.f, 0.f, 0.f, 0.f};
    acc_dk[dt] = {0.f, 0.f, 0.f, 0.f};
  }

  const int q_start = causal ? kv0 : 0;
  for (int q0 = q_start; q0 < T; q0 += QT) {
    // ---- stage Q / dO tiles + lse/delta
    __syncthreads();
    {
      constexpr int CHUNKS = (QT * D) / (256 * 8);
#pragma unroll
      for (int c = 0; c < CHUNKS; ++c) {
        int idx = (c * 256 + threadIdx.x) * 8;
        int row = idx / D, col = idx % D;
        *reinterpret_cast<bf16x8*>(&lds.qt[row][col]) =
            *reinterpret_cast<const bf16x8*>(
                Qh + (long)(q0 + row) * qs.st + col);
        *reinterpret_cast<bf16x8*>(&lds.dot[row][col]) =
            *reinterpret_cast<const bf16x8*>(
                dOh + (long)(q0 + row) * dos.st + col);
      }
      if (threadIdx.x < QT) {
        lds.lse_t[threadIdx.x] = lse_h[q0 + threadIdx.x];
        lds.delta_t[threadIdx.x] = del_h[q0 + threadIdx.x];
      }
    }
    __syncthreads();

    // ---- per 16-q subtile: St, P^T, dPt, dS^T
#pragma unroll
    for (int n = 0; n < 2; ++n) {
      f32x4 st = {0.f, 0.f, 0.f, 0.f};
      f32x4 dpt = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int ds = 0; ds < D / 32; ++ds) {
        bf16x8 ak = *reinterpret_cast<const bf16x8*>(
            &lds.kt[wid * 16 + r][ds * 32 + qg * 8]);
        bf16x8 av;
        if constexpr (HASV)
          av = *reinterpret_cast<const bf16x8*>(
              &lds.vt[wid * 16 + r][ds * 32 + qg * 8]);
        else
          av = *reinterpret_cast<const bf16x8*>(
              Vh + (long)(kv0 + wid * 16 + r) * vs.st + ds * 32 + qg * 8);
        bf16x8 bq = *reinterpret_cast<const bf16x8*>(
            &lds.qt[n * 16 + r][ds * 32 + qg * 8]);
        bf16x8 bdo = *reinterpret_cast<const bf16x8*>(
            &lds.dot[n * 16 + r][ds * 32 + qg * 8]);
        st = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ak, bq, st, 0, 0, 0);
        dpt = __builtin_amdgcn_mfma_f32_16x16x32_bf16(av, bdo, dpt, 0, 0, 0);
      }
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int key_loc = qg * 4 + reg;           // within wave's 16 keys
        const int key_glob = kv0 + wid * 16 + key_loc;
        const int q_glob = q0 + n * 16 + r;
        float p = 0.f;
        if ((!causal || q_glob >= key_glob) && key_glob < kv_len) {
          p = __expf(scale * st[reg] - lds.lse_t[n * 16 + r]);
        }
        const float dsv = p * (dpt[reg] - lds.delta_t[n * 16 + r]) * scale;
        lds.pt[wid][key_loc][n * 16 + r] = f2us(p);
        lds.dst[wid * 16 + key_loc][n * 16 + r] = f2us(dsv);
      }
    }

    // ---- accumulate dV, dK over this q tile (contraction over 32 q)
    {
      bf16x8 ap = *reinterpret_cast<const bf16x8*>(&lds.pt[wid][r][qg * 8]);
      bf16x8 adst =
          *reinterpret_cast<const bf16x8*>(&lds.dst[wid * 16 + r][qg * 8]);
      // Two B-fragment paths (host picks per SAMD_ATTN_BWD_TR16): scalar
      // gathers let hipcc schedule with counted waits across the pt/dst
      // stores; the tr16 hardware-transpose reads are denser but their asm
      // clobber is a scheduling fence (measured both ways — see profiles/).
#pragma unroll
      for (int dt = 0; dt < D / 16; ++dt) {
        bf16x8 bdo, bq;
        if constexpr (TR16) {
          tr16_bfrag2<D + 8, D + 8>(&lds.dot[qg * 8][dt * 16],
                                    &lds.qt[qg * 8][dt * 16], lane, bdo, bq);
        } else {
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            bdo[j] = (short)lds.dot[qg * 8 + j][dt * 16 + r];
            bq[j] = (short)lds.qt[qg * 8 + j][dt * 16 + r];
          }
        }
        acc_dv[dt] =
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(ap, bdo, acc_dv[dt], 0, 0, 0);
        acc_dk[dt] =
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(adst, bq, acc_dk[dt], 0, 0, 0);
      }
    }

    // (dQ moved to attn_dq_kernel — q-block grid, register accumulation,
    // no atomics; this kernel now only accumulates dK/dV.)
    __syncthreads();
  }

  // ---- epilogue: write dK, dV.  With GQA (several q heads per kv head)
  // the partials accumulate into fp32 buffers with atomics; otherwise a
  // direct bf16 store.
  if (n_kv != n_heads) {
    float* dKh = dKf + b * dks.sb + h_kv * dks.sh;
    float* dVh = dVf + b * dvs.sb + h_kv * dvs.sh;
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int key_glob = kv0 + wid * 16 + qg * 4 + reg;
#pragma unroll
      for (int dt = 0; dt < D / 16; ++dt) {
        atomicAdd(&dKh[(long)key_glob * dks.st + dt * 16 + r],
                  acc_dk[dt][reg]);
        atomicAdd(&dVh[(long)key_glob * dvs.st + dt * 16 + r],
                  acc_dv[dt][reg]);
      }
    }
  } else {
    unsigned short* dKh = dK + b * dks.sb + h * dks.sh;
    unsigned short* dVh = dV + b * dvs.sb + h * dvs.sh;
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int key_glob = kv0 + wid * 16 + qg * 4 + reg;
#pragma unroll
      for (int dt = 0; dt < D / 16; ++dt) {
        dKh[(long)key_glob * dks.st + dt * 16 + r] = f2us(acc_dk[dt][reg]);
        dVh[(long)key_glob * dvs.st + dt * 16 + r] = f2us(acc_dv[dt][reg]);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// dQ kernel (backward v2, round 2).  The combined kernel's dQ phase was
// 41-54% of backward time, and its fp32 atomics alone 23-37% (measured via
// the dqmode probe; every dQ element was atomically added by T/KB
// kv-blocks).  This kernel re-parallelizes dQ over Q-BLOCKS: each block
// owns 64 q rows, loops over the key tiles, recomputes S^T and dP^T
// (+2 GEMM units of recompute vs the combined kernel's shared tiles), and
// accumulates dQ in registers — no atomics, no fp32 scratch buffer, no
// bf16 cast pass; the result is stored once as bf16 in the [B,T,H,D]
// physical layout the upstream projections use.
//
// Routing: S^T / dS^T land in the MFMA D-layout with key on the reg dim
// and q on l&15.  The dQ MFMA wants A[m=q][k=key over 32] — exactly the
// cross-quarter exchange validated in tools/bwd_route_probe.hip: pack the
// 4 f32 regs as 2 u32 of bf16 pairs, one __shfl at lane stride 16 per
// pair, assemble 8 slots.  No LDS round-trip for dS^T.
// ---------------------------------------------------------------------------
namespace dq {
constexpr int QB = 64;  // q rows per block (4 waves x 16)
constexpr int KT = 32;  // keys per tile iteration
}  // namespace dq

template <int D, bool GKV>
struct AttnDqLds {
  // GKV (D=256): K/V/Q/dO fragments read straight from global (L2) and
  // only the K tile is LDS-staged for the transposed dQ B-fragment —
  // 97.7 KB -> 17.4 KB of LDS lifts occupancy from 1 to 3 blocks/CU.
  unsigned short qt[GKV ? 1 : dq::QB][D + 8];
  unsigned short dot[GKV ? 1 : dq::QB][D + 8];
  unsigned short kt[dq::KT][D + 8];
  unsigned short vt[GKV ? 1 : dq::KT][D + 8];
  float lse_t[dq::QB];
  float delta_t[dq::QB];
};

template <int D, bool TR16, bool GKV>
__launch_bounds__(256, GKV ? 3 : (D <= 128 ? 2 : 1))
__global__ void attn_dq_kernel(const unsigned short* __restrict__ dO,
                               const unsigned short* __restrict__ Q,
                               const unsigned short* __restrict__ K,
                               const unsigned short* __restrict__ V,
                               const float* __restrict__ LSE,
                               const float* __restrict__ DELTA,
                               unsigned short* __restrict__ dQ16, int T,
                               int n_heads, int n_kv, float scale, int causal,
                               int kv_len, TStr dos, TStr qs, TStr ks,
                               TStr vs, TStr dqs) {
  using namespace dq;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  AttnDqLds<D, GKV>& lds = *reinterpret_cast<AttnDqLds<D, GKV>*>(smem);

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int r = lane & 15;
  const int qg = lane >> 4;
  const int q0 = blockIdx.y * QB;
  const long bh = blockIdx.x;
  const long b = bh / n_heads, h = bh % n_heads;
  const long h_kv = h * n_kv / n_heads;
  const unsigned short* Qh = Q + b * qs.sb + h * qs.sh;
  const unsigned short* Kh = K + b * ks.sb + h_kv * ks.sh;
  const unsigned short* Vh = V + b * vs.sb + h_kv * vs.sh;
  const unsigned short* dOh = dO + b * dos.sb + h * dos.sh;
  const float* lse_h = LSE + bh * (long)T;
  const float* del_h = DELTA + bh * (long)T;

  // ---- stage the block's Q / dO rows + lse/delta once (GKV reads the
  // row fragments straight from L2 instead)
  {
    if constexpr (!GKV) {
      constexpr int CHUNKS = (QB * D) / (256 * 8);
#pragma unroll
      for (int c = 0; c < CHUNKS; ++c) {
        int idx = (c * 256 + threadIdx.x) * 8;
        int row = idx / D, col = idx % D;
        *reinterpret_cast<bf16x8*>(&lds.qt[row][col]) =
            *reinterpret_cast<const bf16x8*>(Qh + (long)(q0 + row) * qs.st + col);
        *reinterpret_cast<bf16x8*>(&lds.dot[row][col]) =
            *reinterpret_cast<const bf16x8*>(
                dOh + (long)(q0 + row) * dos.st + col);
      }
    }
    if (threadIdx.x < QB) {
      lds.lse_t[threadIdx.x] = lse_h[q0 + threadIdx.x];
      lds.delta_t[threadIdx.x] = del_h[q0 + threadIdx.x];
    }
  }

  f32x4 acc[D / 16];
#pragma unroll
  for (int dt = 0; dt < D / 16; ++dt) acc[dt] = {0.f, 0.f, 0.f, 0.f};

  const int k_end = causal ? (q0 + QB < T ? q0 + QB : T) : T;
  for (int k0 = 0; k0 < k_end; k0 += KT) {
    __syncthreads();  // previous tile fully consumed (also orders staging)
    {
      constexpr int CHUNKS = (KT * D) / (256 * 8);
#pragma unroll
      for (int c = 0; c < CHUNKS; ++c) {
        int idx = (c * 256 + threadIdx.x) * 8;
        int row = idx / D, col = idx % D;
        *reinterpret_cast<bf16x8*>(&lds.kt[row][col]) =
            *reinterpret_cast<const bf16x8*>(
                Kh + (long)(k0 + row) * ks.st + col);
        if constexpr (!GKV)
          *reinterpret_cast<bf16x8*>(&lds.vt[row][col]) =
              *reinterpret_cast<const bf16x8*>(
                  Vh + (long)(k0 + row) * vs.st + col);
      }
    }
    __syncthreads();

    // Waves whose 16 q rows are entirely left of this key tile contribute
    // nothing (causal) — skip the math, keep the barriers block-uniform.
    if (causal && k0 > q0 + wid * 16 + 15) continue;

    // ---- per 16-key subtile: S^T, dP^T -> dS^T (key on reg, q on lane)
    float ds_sub[2][4];
#pragma unroll
    for (int n = 0; n < 2; ++n) {
      f32x4 st = {0.f, 0.f, 0.f, 0.f};
      f32x4 dpt = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int ds_ = 0; ds_ < D / 32; ++ds_) {
        bf16x8 ak = *reinterpret_cast<const bf16x8*>(
            &lds.kt[n * 16 + r][ds_ * 32 + qg * 8]);
        bf16x8 av, bq, bdo;
        if constexpr (GKV) {
          av = *reinterpret_cast<const bf16x8*>(
              Vh + (long)(k0 + n * 16 + r) * vs.st + ds_ * 32 + qg * 8);
          bq = *reinterpret_cast<const bf16x8*>(
              Qh + (long)(q0 + wid * 16 + r) * qs.st + ds_ * 32 + qg * 8);
          bdo = *reinterpret_cast<const bf16x8*>(
              dOh + (long)(q0 + wid * 16 + r) * dos.st + ds_ * 32 + qg * 8);
        } else {
          av = *reinterpret_cast<const bf16x8*>(
              &lds.vt[n * 16 + r][ds_ * 32 + qg * 8]);
          bq = *reinterpret_cast<const bf16x8*>(
              &lds.qt[wid * 16 + r][ds_ * 32 + qg * 8]);
          bdo = *reinterpret_cast<const bf16x8*>(
              &lds.dot[wid * 16 + r][ds_ * 32 + qg * 8]);
        }
        st = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ak, bq, st, 0, 0, 0);
        dpt = __builtin_amdgcn_mfma_f32_16x16x32_bf16(av, bdo, dpt, 0, 0, 0);
      }
      const int q_glob = q0 + wid * 16 + r;
      const float l_q = lds.lse_t[wid * 16 + r];
      const float d_q = lds.delta_t[wid * 16 + r];
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int key_glob = k0 + n * 16 + qg * 4 + reg;
        float p = 0.f;
        if ((!causal || q_glob >= key_glob) && key_glob < kv_len)
          p = __expf(scale * st[reg] - l_q);
        ds_sub[n][reg] = p * (dpt[reg] - d_q) * scale;
      }
    }

    // ---- route dS^T D-layout -> A[m=q][k=key 0..31] (bwd_route_probe)
    unsigned pk[2][2];
#pragma unroll
    for (int n = 0; n < 2; ++n) {
      pk[n][0] = (unsigned)f2us(ds_sub[n][0]) |
                 ((unsigned)f2us(ds_sub[n][1]) << 16);
      pk[n][1] = (unsigned)f2us(ds_sub[n][2]) |
                 ((unsigned)f2us(ds_sub[n][3]) << 16);
    }
    const int a = (qg & 1) * 2;   // first source quarter
    const int nsrc = qg >> 1;     // source subtile (keys 0-15 / 16-31)
    const int s0 = a * 16 + r;
    const int s1 = (a + 1) * 16 + r;
    // shuffle BOTH subtiles and select with the destination's nsrc after:
    // __shfl evaluates its operand in the SOURCE lane, whose nsrc differs
    unsigned rr[4][2];
#pragma unroll
    for (int nn = 0; nn < 2; ++nn) {
      rr[nn * 2 + 0][0] = (unsigned)__shfl((int)pk[nn][0], s0, 64);
      rr[nn * 2 + 0][1] = (unsigned)__shfl((int)pk[nn][1], s0, 64);
      rr[nn * 2 + 1][0] = (unsigned)__shfl((int)pk[nn][0], s1, 64);
      rr[nn * 2 + 1][1] = (unsigned)__shfl((int)pk[nn][1], s1, 64);
    }
    bf16x8 ads;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int src = (j >= 4) ? 1 : 0;
      const int reg = j & 3;
      const unsigned u = rr[nsrc * 2 + src][reg >> 1];
      ads[j] = (short)((reg & 1) ? (u >> 16) : (u & 0xffffu));
    }

    // ---- dQ += dS @ K  (B-fragment: K transposed from the kt tile)
#pragma unroll
    for (int dt = 0; dt < D / 16; ++dt) {
      bf16x8 bk;
      if constexpr (TR16) {
        bk = tr16_bfrag<D + 8>(&lds.kt[qg * 8][dt * 16], lane);
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j)
          bk[j] = (short)lds.kt[qg * 8 + j][dt * 16 + r];
      }
      acc[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ads, bk, acc[dt], 0, 0, 0);
    }
  }

  // ---- store dQ once, bf16, [B,T,H,D]-physical strides
  unsigned short* dQh = dQ16 + b * dqs.sb + h * dqs.sh;
#pragma unroll
  for (int reg = 0; reg < 4; ++reg) {
    const int q_glob = q0 + wid * 16 + qg * 4 + reg;
#pragma unroll
    for (int dt = 0; dt < D / 16; ++dt)
      dQh[(long)q_glob * dqs.st + dt * 16 + r] = f2us(acc[dt][reg]);
  }
}

std::vector<at::Tensor> attn_bwd(at::Tensor dout, at::Tensor q, at::Tensor k,
                                 at::Tensor v, at::Tensor o, at::Tensor lse,
                                 bool causal, int64_t kv_len) {
  using namespace bwd;
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == at::kBFloat16);
  const int B = (int)q.size(0), H = (int)q.size(1), T = (int)q.size(2),
            D = (int)q.size(3);
  const int Hkv = (int)k.size(1);
  TORCH_CHECK(H % Hkv == 0);
  TORCH_CHECK(T % KB == 0, "attn_bwd: T must be a multiple of 64");
  if (kv_len <= 0) kv_len = T;
  TORCH_CHECK(kv_len <= T, "attn_bwd: kv_len exceeds padded T");
  TORCH_CHECK(D == 64 || D == 128 || D == 256);
  TORCH_CHECK(dout.stride(3) == 1 && q.stride(3) == 1 && k.stride(3) == 1 &&
              v.stride(3) == 1, "attn_bwd: innermost dim must be dense");
  TORCH_CHECK(dout.stride(2) % 8 == 0 && q.stride(2) % 8 == 0 &&
              k.stride(2) % 8 == 0 && v.stride(2) % 8 == 0);

  auto delta = at::empty({B, H, T}, q.options().dtype(at::kFloat));
  // grads physically [B, T, H, D] (matches the projection layout upstream,
  // so the model-side transposes stay views).  dQ is written exactly once
  // by the q-block dQ kernel: bf16 direct, no fp32 scratch, no zero-fill.
  auto dq = at::empty({B, T, H, D}, q.options()).permute({0, 2, 1, 3});
  const bool gqa = Hkv != H;
  auto dk = at::empty({B, T, Hkv, D}, k.options()).permute({0, 2, 1, 3});
  auto dv = at::empty({B, T, Hkv, D}, v.options()).permute({0, 2, 1, 3});
  at::Tensor dk_f32, dv_f32;
  if (gqa) {
    dk_f32 = at::zeros({B, T, Hkv, D}, q.options().dtype(at::kFloat))
                 .permute({0, 2, 1, 3});
    dv_f32 = at::zeros({B, T, Hkv, D}, q.options().dtype(at::kFloat))
                 .permute({0, 2, 1, 3});
  }
  auto stream = at::hip::getCurrentHIPStream();
  const float scale = 1.f / sqrtf((float)D);
  dim3 grid(B * H, T / KB), block(256);
  // default ON for D<=128 (measured +2-11% bwd across shapes; numerics
  // validated both ways); SAMD_ATTN_BWD_TR16=0 reverts to scalar gathers
  static const bool use_tr16 = [] {
    const char* e = getenv("SAMD_ATTN_BWD_TR16");
    return !(e && e[0] == '0');
  }();
  static const bool use_dq_gkv = [] {
    const char* e = getenv("SAMD_ATTN_DQ_GKV");
    return e && e[0] == '1';
  }();
  // V tile from L2 instead of LDS at D<=128 (SAMD_ATTN_BWD_VL2=1): trades
  // per-q-tile L2 re-reads for 62->45 KB LDS/block (3 blocks/CU scalar path)
  static const bool use_vl2 = [] {
    const char* e = getenv("SAMD_ATTN_BWD_VL2");
    return e && e[0] == '1';
  }();

#define BWD_ARGS                                                             \
    stream.stream(),                                                         \
    reinterpret_cast<const unsigned short*>(dout.data_ptr()),                \
    reinterpret_cast<const unsigned short*>(q.data_ptr()),                   \
    reinterpret_cast<const unsigned short*>(k.data_ptr()),                   \
    reinterpret_cast<const unsigned short*>(v.data_ptr()),                   \
    lse.data_ptr<float>(), delta.data_ptr<float>(),                          \
    reinterpret_cast<unsigned short*>(dk.data_ptr()),                        \
    reinterpret_cast<unsigned short*>(dv.data_ptr()),                        \
    gqa ? dk_f32.data_ptr<float>() : nullptr,                                \
    gqa ? dv_f32.data_ptr<float>() : nullptr, T, H, Hkv,                     \
    scale, causal ? 1 : 0, (int)kv_len, str_of(dout), str_of(q),             \
    str_of(k), str_of(v),                                                    \
    gqa ? str_of(dk_f32) : str_of(dk),                                       \
    gqa ? str_of(dv_f32) : str_of(dv)

#define LAUNCH_B(DD)                                                         \
  do {                                                                       \
    dim3 dgrid(std::min((T + 3) / 4, 2048), B * H), dblock(256);             \
    hipLaunchKernelGGL((attn_delta_kernel<DD>), dgrid, dblock, 0,            \
                       stream.stream(),                                      \
                       reinterpret_cast<const unsigned short*>(dout.data_ptr()), \
                       reinterpret_cast<const unsigned short*>(o.data_ptr()), \
                       delta.data_ptr<float>(), T, H, str_of(dout),          \
                       str_of(o));                                           \
    /* GKV (K/V/Q/dO fragments from L2, 17 KB LDS, 3 blocks/CU) measured \
       SLOWER at D=256 (612 vs 525 us at the GPT-J bench shape: the L2   \
       re-reads cost more than the occupancy gains) — kept as an opt-in  \
       probe only */                                                       \
    const bool gkv = use_dq_gkv && DD > 128;                                 \
    size_t SH_DQ = gkv ? sizeof(AttnDqLds<DD, true>)                         \
                       : sizeof(AttnDqLds<DD, false>);                       \
    static bool dq_attr_##DD = [] {                                          \
      hipFuncSetAttribute(                                                   \
          reinterpret_cast<const void*>(&attn_dq_kernel<DD, false, false>),  \
          hipFuncAttributeMaxDynamicSharedMemorySize,                        \
          (int)sizeof(AttnDqLds<DD, false>));                                \
      hipFuncSetAttribute(                                                   \
          reinterpret_cast<const void*>(&attn_dq_kernel<DD, true, false>),   \
          hipFuncAttributeMaxDynamicSharedMemorySize,                        \
          (int)sizeof(AttnDqLds<DD, false>));                                \
      return true;                                                           \
    }();                                                                     \
    (void)dq_attr_##DD;                                                      \
    {                                                                        \
      dim3 qgrid(B * H, T / dq::QB), qblock(256);                            \
      if (gkv)                                                               \
        hipLaunchKernelGGL((attn_dq_kernel<DD, false, true>), qgrid,         \
                           qblock, SH_DQ, stream.stream(),                   \
                           reinterpret_cast<const unsigned short*>(dout.data_ptr()), \
                           reinterpret_cast<const unsigned short*>(q.data_ptr()), \
                           reinterpret_cast<const unsigned short*>(k.data_ptr()), \
                           reinterpret_cast<const unsigned short*>(v.data_ptr()), \
                           lse.data_ptr<float>(), delta.data_ptr<float>(),   \
                           reinterpret_cast<unsigned short*>(dq.data_ptr()), \
                           T, H, Hkv, scale, causal ? 1 : 0, (int)kv_len,   \
                           str_of(dout), str_of(q), str_of(k), str_of(v),    \
                           str_of(dq));                                      \
      else if (use_tr16 && DD <= 128)                                        \
        hipLaunchKernelGGL((attn_dq_kernel<DD, true, false>), qgrid,         \
                           qblock, SH_DQ, stream.stream(),                   \
                           reinterpret_cast<const unsigned short*>(dout.data_ptr()), \
                           reinterpret_cast<const unsigned short*>(q.data_ptr()), \
                           reinterpret_cast<const unsigned short*>(k.data_ptr()), \
                           reinterpret_cast<const unsigned short*>(v.data_ptr()), \
                           lse.data_ptr<float>(), delta.data_ptr<float>(),   \
                           reinterpret_cast<unsigned short*>(dq.data_ptr()), \
                           T, H, Hkv, scale, causal ? 1 : 0, (int)kv_len,   \
                           str_of(dout), str_of(q), str_of(k), str_of(v),    \
                           str_of(dq));                                      \
      else                                                                   \
        hipLaunchKernelGGL((attn_dq_kernel<DD, false, false>), qgrid,        \
                           qblock, SH_DQ, stream.stream(),                   \
                           reinterpret_cast<const unsigned short*>(dout.data_ptr()), \
                           reinterpret_cast<const unsigned short*>(q.data_ptr()), \
                           reinterpret_cast<const unsigned short*>(k.data_ptr()), \
                           reinterpret_cast<const unsigned short*>(v.data_ptr()), \
                           lse.data_ptr<float>(), delta.data_ptr<float>(),   \
                           reinterpret_cast<unsigned short*>(dq.data_ptr()), \
                           T, H, Hkv, scale, causal ? 1 : 0, (int)kv_len,   \
                           str_of(dout), str_of(q), str_of(k), str_of(v),    \
                           str_of(dq));                                      \
    }                                                                        \
    const bool tr = use_tr16 && DD <= 128;                                   \
    const bool vl2 = use_vl2 && DD <= 128;                                   \
    constexpr size_t SH_V = sizeof(AttnBwdLds<DD, (DD <= 128)>);             \
    constexpr size_t SH_N = sizeof(AttnBwdLds<DD, false>);                   \
    size_t shmem = vl2 ? SH_N : SH_V;                                        \
    static bool attr_set_##DD = [] {                                         \
      hipFuncSetAttribute(                                                   \
          reinterpret_cast<const void*>(&attn_bwd_kernel<DD, false, false>), \
          hipFuncAttributeMaxDynamicSharedMemorySize, (int)SH_V);            \
      hipFuncSetAttribute(                                                   \
          reinterpret_cast<const void*>(&attn_bwd_kernel<DD, true, false>),  \
          hipFuncAttributeMaxDynamicSharedMemorySize, (int)SH_V);            \
      hipFuncSetAttribute(                                                   \
          reinterpret_cast<const void*>(&attn_bwd_kernel<DD, false, true>),  \
          hipFuncAttributeMaxDynamicSharedMemorySize, (int)SH_N);            \
      hipFuncSetAttribute(                                                   \
          reinterpret_cast<const void*>(&attn_bwd_kernel<DD, true, true>),   \
          hipFuncAttributeMaxDynamicSharedMemorySize, (int)SH_N);            \
      return true;                                                           \
    }();                                                                     \
    (void)attr_set_##DD;                                                     \
    if (tr && vl2)                                                           \
      hipLaunchKernelGGL((attn_bwd_kernel<DD, true, true>), grid, block,     \
                         shmem, BWD_ARGS);                                   \
    else if (tr)                                                             \
      hipLaunchKernelGGL((attn_bwd_kernel<DD, true, false>), grid, block,    \
                         shmem, BWD_ARGS);                                   \
    else if (vl2)                                                            \
      hipLaunchKernelGGL((attn_bwd_kernel<DD, false, true>), grid, block,    \
                         shmem, BWD_ARGS);                                   \
    else                                                                     \
      hipLaunchKernelGGL((attn_bwd_kernel<DD, false, false>), grid, block,   \
                         shmem, BWD_ARGS);                                   \
  } while (0)

  if (D == 64) LAUNCH_B(64);
  else if (D == 128) LAUNCH_B(128);
  else LAUNCH_B(256);
#undef LAUNCH_B
  if (gqa) return {dq, dk_f32.to(at::kBFloat16), dv_f32.to(at::kBFloat16)};
  return {dq, dk, dv};
}

}  // namespace samd

