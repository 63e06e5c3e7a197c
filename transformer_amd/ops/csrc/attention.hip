// Fused flash-style multi-head attention, forward + backward (SURVEY.md
// K2-K5 + K14; semantics = reference Attention.py:3-34 incl. the additive
// mask*(-1e9) before softmax, reference Attention.py:25-26).
//
// Layout: q (B,Sq,H,dh), k/v (B,Sk,H,dh) bf16 — the (B,S,H,dh) layout the
// packed QKV GEMM produces, so the reference's head split/merge transposes
// (Attention.py:52-57,74-76) never exist on device.
// Masking: `causal` flag + per-token kv_pad bytes replace the materialized
// (B,1,1,S)/(T,T) mask tensors (K14): three variants — encoder self
// (pad), decoder self (causal+pad), cross (pad, Sq != Sk).
//
// Forward: online softmax (running m, l per row), O(Sq·dh) memory, fp32
// MFMA accumulate, saves LSE for the backward.  Work unit: 4-wave block of
// 64 q-rows (128 for seq>=2048 via the RF=2 variant); 64-key K/V tiles
// staged through swizzled LDS; QK^T and PV on mfma_f32_16x16x32_bf16; P
// crosses C-layout -> A-layout through a small per-wave LDS buffer; the PV
// B-operand is read from the natural V image with ds_read_b64_tr_b16.
//
// Backward: FA2-style two-kernel split (no atomics, deterministic):
//   attn_bwd_kv: grid over kv-tiles, accumulates dK,dV (recomputing P from
//                q,k,lse), dP from dO·V^T, dS = P*(dP-D).
//   attn_bwd_q:  grid over q-tiles, accumulates dQ.
//   attn_bwd_dot: D[row] = rowsum(dO * O).
#include "common.h"

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));

#define KVT 32        // backward kv/q tile (fwd uses FKVT=64)
#define QW 16         // q-rows per wave
#define WAVES 4
#define NEG_BIG (-1e9f)

// ---- swizzled LDS helpers --------------------------------------------------
// natural tile [rows][DH] bf16 (row stride DH*2 bytes): XOR spreads the
// 16-lane b128 groups; transposed/P tiles [rows][32] (64-byte rows).
template <int ROWB>  // row stride in bytes
DEV_INLINE int lds_swz(int row, int col_bytes) {
  if (ROWB >= 128) return row * ROWB + (col_bytes ^ ((row & 7) << 4));
  return row * ROWB + (col_bytes ^ ((row & 3) << 4));
}

template <int ROWB>
DEV_INLINE bf16x8 lds_read8(const short* base, int row, int col /*elems*/) {
  return (bf16x8)*(const s16x8*)((const char*)base + lds_swz<ROWB>(row, col * 2));
}

template <int ROWB>
DEV_INLINE void lds_write8(short* base, int row, int col, s16x8 v) {
  *(s16x8*)((char*)base + lds_swz<ROWB>(row, col * 2)) = v;
}

template <int ROWB>
DEV_INLINE void lds_write1(short* base, int row, int col, short v) {
  *(short*)((char*)base + (row * ROWB + ((col * 2) ^ ((ROWB >= 128 ? (row & 7) : (row & 3)) << 4)))) = v;
}


// Read an MFMA B-fragment COLUMN-wise from a natural [rows][DH] image via
// 2x ds_read_b64_tr_b16 (hardware transpose read, guide T10; lane
// semantics verified by tools/tr16_probe.hip): lane l receives
// T[kb .. kb+8)[col0 + (l&15)].  Replaces the scalar-write transposed
// LDS images (per-element ds_write_b16 scatter) entirely.
typedef short s16x4t __attribute__((ext_vector_type(4)));

template <int ROWB>
DEV_INLINE bf16x8 lds_read8_tr(const short* base, int kb, int col0) {
  const int mp = threadIdx.x & 15;
  const int row = kb + (mp >> 2);
  const int cb = (col0 + 4 * (mp & 3)) * 2;
  s16x4t lo = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
      (__attribute__((address_space(3))) s16x4t*)(
          const_cast<char*>((const char*)base) + lds_swz<ROWB>(row, cb)));
  s16x4t hi = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
      (__attribute__((address_space(3))) s16x4t*)(
          const_cast<char*>((const char*)base) + lds_swz<ROWB>(row + 4, cb)));
  s16x8 v = {lo[0], lo[1], lo[2], lo[3], hi[0], hi[1], hi[2], hi[3]};
  return (bf16x8)v;
}


// Per-wave C-scratch in TRANSPOSED layout [C-col][16 C-rows]: each lane
// writes its 4 accumulator rows (fixed C-col) as ONE b64 store, and the
// next stage reads MFMA A-fragments back with ds_read_b64_tr_b16 — no
// scalar b16 scatter in either direction.
DEV_INLINE void scrT_write4(short* img, int ccol, int rr0, s16x4 v) {
  *(s16x4*)&img[ccol * 16 + rr0] = v;
}

// A-fragment from the transposed scratch: lane l receives rows
// (inner C-row = l&15) x outer chunk [ob, ob+8).
DEV_INLINE bf16x8 scrT_read8(const short* img, int ob) {
  const int mp = threadIdx.x & 15;
  const char* base = (const char*)img;
  s16x4t lo = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
      (__attribute__((address_space(3))) s16x4t*)(
          const_cast<char*>(base) + ((ob + (mp >> 2)) * 16 + 4 * (mp & 3)) * 2));
  s16x4t hi = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
      (__attribute__((address_space(3))) s16x4t*)(
          const_cast<char*>(base) +
          ((ob + 4 + (mp >> 2)) * 16 + 4 * (mp & 3)) * 2));
  s16x8 v = {lo[0], lo[1], lo[2], lo[3], hi[0], hi[1], hi[2], hi[3]};
  return (bf16x8)v;
}

// Stage a [KVT][DH] bf16 tile from global rows (stride row_stride elems)
// into swizzled LDS: natural layout into lds_n (if WRITE_N) and/or the
// transpose [DH][KVT] into lds_t (if WRITE_T).
// 256 threads cooperate; guards rows >= nrows with zeros.
template <int DH, bool WRITE_N, bool WRITE_T, int NR = KVT>
DEV_INLINE void stage_kv(const short* __restrict__ g, long row_stride,
                         int nrows, short* lds_n, short* lds_t) {
  const int t = threadIdx.x;
  constexpr int CH = NR * DH / 8;   // 16-byte chunks
  constexpr int NP = (CH + 255) / 256;
  // issue ALL global loads before any LDS write: at reduced occupancy the
  // staging must keep several loads in flight per lane to reach HBM rate
  s16x8 v[NP];
#pragma unroll
  for (int p = 0; p < NP; ++p) {
    int idx = p * 256 + t;
    int row = idx / (DH / 8);
    int c8 = (idx % (DH / 8)) * 8;
    v[p] = {0, 0, 0, 0, 0, 0, 0, 0};
    if (idx < CH && row < nrows)
      v[p] = *(const s16x8*)(g + row * row_stride + c8);
  }
#pragma unroll
  for (int p = 0; p < NP; ++p) {
    int idx = p * 256 + t;
    if (idx >= CH) break;
    int row = idx / (DH / 8);
    int c8 = (idx % (DH / 8)) * 8;
    if (WRITE_N) lds_write8<DH * 2>(lds_n, row, c8, v[p]);
    if (WRITE_T) {
#pragma unroll
      for (int j = 0; j < 8; ++j)
        lds_write1<NR * 2>(lds_t, c8 + j, row, v[p][j]);
    }
  }
}

// ---- forward ---------------------------------------------------------------
// TRV: read the PV B-operand from the NATURAL V image via
// ds_read_b64_tr_b16 (no transposed staging); else build a transposed V
// image with scalar writes and read it row-wise.  Same buffer either way;
// the faster variant is chosen by measurement (tools/attn_bench.py).
//
// Each wave owns FQW=32 q-rows as RF=2 16-row fragments, so one workgroup
// covers 128 q-rows: at seq 4096 the kernel is bound by re-staging K/V
// once per q-block (16.4 GB per call at 64-row blocks), and doubling the
// block height halves that traffic.  Softmax temporaries live per
// row-fragment iteration; only qf/acc/m/l are duplicated.
template <int DH, bool TRV, int RF>
__global__ __launch_bounds__(256)
void attn_fwd_kernel(const short* __restrict__ Q, const short* __restrict__ K,
                     const short* __restrict__ V,
                     const unsigned char* __restrict__ kv_pad,
                     short* __restrict__ O, float* __restrict__ LSE, int B,
                     int H, int Sq, int Sk, int causal, float scale,
                     long q_rs, long kv_rs, long o_rs, long q_bs, long kv_bs) {
  constexpr int D32 = DH / 32;   // QK^T MFMA k-steps
  constexpr int D16 = DH / 16;   // O fragments
  constexpr int FKVT = 64;       // key tile
  constexpr int NHALF = FKVT / 16;
  constexpr int FQW = 16 * RF;   // q-rows per wave (RF template)
  __shared__ short k_lds[FKVT * DH];
  __shared__ short v_lds[FKVT * DH];   // natural [FKVT][DH] or transposed
  __shared__ short p_lds[WAVES][QW * FKVT];  // one 16-row rf at a time

  const int bh = blockIdx.x;      // b*H + h
  const int qb = blockIdx.y;      // q-block of WAVES*FQW rows
  const int b = bh / H, h = bh % H;
  const int wid = threadIdx.x >> 6, lane = threadIdx.x & 63;
  const int q0 = qb * (WAVES * FQW) + wid * FQW;  // wave's first q-row
  const int fr = lane & 15, kg = lane >> 4;

  const short* Qp = Q + b * q_bs + (long)h * DH;
  const short* Kp = K + b * kv_bs + (long)h * DH;
  const short* Vp = V + b * kv_bs + (long)h * DH;
  const unsigned char* pad = kv_pad ? kv_pad + (long)b * Sk : nullptr;

  // Q fragments in registers: A[row=fr][k=kg*8+j] per 32-chunk, per rf
  bf16x8 qf[RF][D32];
#pragma unroll
  for (int rf = 0; rf < RF; ++rf)
#pragma unroll
    for (int d = 0; d < D32; ++d) {
      s16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
      int row = q0 + rf * 16 + fr;
      if (row < Sq)
        v = *(const s16x8*)(Qp + (long)row * q_rs + d * 32 + kg * 8);
      qf[rf][d] = (bf16x8)v;
    }

  f32x4 acc[RF][D16];
#pragma unroll
  for (int rf = 0; rf < RF; ++rf)
#pragma unroll
    for (int i = 0; i < D16; ++i) acc[rf][i] = {0, 0, 0, 0};
  float m_run[RF][4], l_run[RF][4];
#pragma unroll
  for (int rf = 0; rf < RF; ++rf)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      m_run[rf][r] = -1e30f;
      l_run[rf][r] = 0.f;
    }

  // causal: keys beyond this block's last row are fully masked
  const int kend = causal ? min(Sk, (qb + 1) * (WAVES * FQW)) : Sk;

  for (int k0 = 0; k0 < kend; k0 += FKVT) {
    const int kc = min(FKVT, Sk - k0);
    stage_kv<DH, true, false, FKVT>(Kp + (long)k0 * kv_rs, kv_rs, kc,
                                    k_lds, nullptr);
    if (TRV)
      stage_kv<DH, true, false, FKVT>(Vp + (long)k0 * kv_rs, kv_rs, kc,
                                      v_lds, nullptr);
    else
      stage_kv<DH, false, true, FKVT>(Vp + (long)k0 * kv_rs, kv_rs, kc,
                                      nullptr, v_lds);
    __syncthreads();
    // tile-uniform mask facts: most tiles (all of them when Sk%64==0 and
    // nothing is padded; all-but-diagonal ones under causal) need NO
    // per-element mask work — the kernel is issue-bound, and the mask
    // loop is ~76 VALU per 64-key tile per row-fragment.
    bool tile_pad_any = false;
    if (pad) {
      unsigned char pb =
          (lane < FKVT && k0 + lane < Sk) ? pad[k0 + lane] : 0;
      tile_pad_any = __any(pb != 0);
    }
    const bool tile_full = (kc == FKVT) && !tile_pad_any;

#pragma unroll
    for (int rf = 0; rf < RF; ++rf) {
      const int qr0 = q0 + rf * 16;
      // ---- S = scale*(Q K^T) + mask, NHALF 16-key halves -------------
      float p_raw[NHALF][4];
      float tile_pmax[4];
#pragma unroll
      for (int r = 0; r < 4; ++r) tile_pmax[r] = -1e30f;
      // all NHALF QK^T chains issued back-to-back (independent
      // accumulators -> MFMA ILP) before any softmax VALU
      f32x4 s2h[NHALF];
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int half = 0; half < NHALF; ++half) {
        s2h[half] = {0, 0, 0, 0};
#pragma unroll
        for (int d = 0; d < D32; ++d) {
          bf16x8 kf =
              lds_read8<DH * 2>(k_lds, half * 16 + fr, d * 32 + kg * 8);
          s2h[half] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              qf[rf][d], kf, s2h[half], 0, 0, 0);
        }
      }
      __builtin_amdgcn_s_setprio(0);
      // clean: whole tile unmasked for this wave's rows (wave-uniform)
      const bool clean =
          tile_full && (!causal || k0 + FKVT - 1 <= qr0 + kg * 4);
      if (clean) {
#pragma unroll
        for (int half = 0; half < NHALF; ++half)
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            float x = s2h[half][r] * scale;
            p_raw[half][r] = x;
            tile_pmax[r] = fmaxf(tile_pmax[r], x);
          }
      } else {
#pragma unroll
        for (int half = 0; half < NHALF; ++half) {
          const int kcol = k0 + half * 16 + fr;   // C col = lane&15
          const bool col_pad = (kcol >= Sk) || (pad && pad[min(kcol, Sk - 1)]);
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const int qrow = qr0 + kg * 4 + r;    // C row = (lane>>4)*4+r
            float x = s2h[half][r] * scale;
            if (col_pad) x += NEG_BIG;
            if (causal && kcol > qrow) x += NEG_BIG;
            p_raw[half][r] = x;
            tile_pmax[r] = fmaxf(tile_pmax[r], x);
          }
        }
      }
      // row max across the 16 cols held by the 16-lane group
#pragma unroll
      for (int r = 0; r < 4; ++r) {
#pragma unroll
        for (int off = 1; off < 16; off <<= 1)
          tile_pmax[r] = fmaxf(tile_pmax[r], __shfl_xor(tile_pmax[r], off));
      }
      // defer-max (guide T13, THR=8): when no row's max grew past m+THR,
      // keep the old max (P bounded by e^THR, fp32 l/acc absorb it, the
      // final O/l and LSE are unchanged) and skip the alpha exps + the
      // O-wide rescale.  The decision covers this tile's P entirely
      // (previous tile's PV is complete) — the T13 hazard order holds.
      float need = 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r)
        need = fmaxf(need, tile_pmax[r] - m_run[rf][r]);
      float alpha[4];
      if (!__all(need <= 8.0f)) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          float mn = fmaxf(m_run[rf][r], tile_pmax[r]);
          alpha[r] = __expf(m_run[rf][r] - mn);
          m_run[rf][r] = mn;
        }
#pragma unroll
        for (int i = 0; i < D16; ++i)
#pragma unroll
          for (int r = 0; r < 4; ++r) acc[rf][i][r] *= alpha[r];
      } else {
#pragma unroll
        for (int r = 0; r < 4; ++r) alpha[r] = 1.0f;
      }

      // P = exp(S - m), park bf16 P in LDS (A-layout).  The row-sum for l
      // is NOT shuffle-reduced here: it comes out of the PV phase below as
      // one extra MFMA per 32-key chunk against a ones B-operand (the
      // per-half 16-lane shuffle chains were ~160 issue slots per tile in
      // an issue-bound kernel, ~2 MFMAs are ~34 cycles).  l then also
      // normalizes by the sum of the SAME bf16-rounded P the O
      // accumulation uses.
#pragma unroll
      for (int half = 0; half < NHALF; ++half) {
#pragma unroll
        for (int r = 0; r < 4; ++r)
          p_raw[half][r] = __expf(p_raw[half][r] - m_run[rf][r]);
        // one b64 store: P rows q=kg*4.. at transposed-scratch col = key
        s16x4 pw = {f2bfbits(p_raw[half][0]), f2bfbits(p_raw[half][1]),
                    f2bfbits(p_raw[half][2]), f2bfbits(p_raw[half][3])};
        scrT_write4(p_lds[wid], half * 16 + fr, kg * 4, pw);
      }
      // p_lds is per-wave: a wave barrier orders the P writes against
      // this wave's own PV reads.
      __builtin_amdgcn_wave_barrier();

      // ---- O += P · V  (A = P[q][key] from LDS, B = V^T[d][key]) -----
      const short one_bits = (short)0x3F80;  // bf16 1.0
      const bf16x8 onesv = (bf16x8)(s16x8){one_bits, one_bits, one_bits,
                                           one_bits, one_bits, one_bits,
                                           one_bits, one_bits};
      f32x4 accl = {0.f, 0.f, 0.f, 0.f};
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int ks = 0; ks < FKVT / 32; ++ks) {
        bf16x8 pa = scrT_read8(p_lds[wid], ks * 32 + kg * 8);
        accl = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pa, onesv, accl,
                                                       0, 0, 0);
#pragma unroll
        for (int i = 0; i < D16; ++i) {
          bf16x8 vb = TRV
              ? lds_read8_tr<DH * 2>(v_lds, ks * 32 + kg * 8, i * 16)
              : lds_read8<FKVT * 2>(v_lds, i * 16 + fr, ks * 32 + kg * 8);
          acc[rf][i] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              pa, vb, acc[rf][i], 0, 0, 0);
        }
      }
      __builtin_amdgcn_s_setprio(0);
#pragma unroll
      for (int r = 0; r < 4; ++r)
        l_run[rf][r] = l_run[rf][r] * alpha[r] + accl[r];
      __builtin_amdgcn_wave_barrier();  // p_lds reused by next rf
    }
    __syncthreads();
  }

  // epilogue: O = acc / l ; LSE = m + log(l)
  short* Op = O + b * (Sq * o_rs) + (long)h * DH;
#pragma unroll
  for (int rf = 0; rf < RF; ++rf) {
#pragma unroll
    for (int i = 0; i < D16; ++i) {
      const int gcol = i * 16 + fr;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qrow = q0 + rf * 16 + kg * 4 + r;
        if (qrow >= Sq) continue;
        float l = l_run[rf][r];
        float o = (l > 0.f) ? acc[rf][i][r] / l : 0.f;
        Op[(long)qrow * o_rs + gcol] = f2bfbits(o);
      }
    }
    if (fr == 0) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qrow = q0 + rf * 16 + kg * 4 + r;
        if (qrow < Sq)
          LSE[((long)b * H + h) * Sq + qrow] =
              m_run[rf][r] + __logf(fmaxf(l_run[rf][r], 1e-30f));
      }
    }
  }
}

// ---- backward helpers ------------------------------------------------------

// D[row] = sum_d dO[row][d] * O[row][d]  (fp32).  DH/8 threads per row,
// each loading ONE contiguous s16x8 chunk, so a wave's loads are fully
// coalesced (the one-thread-per-row version strided lanes 2*DH bytes
// apart — 12.5% bus efficiency at DH=64, measured 6x off roofline; the
// round-1 wave-per-row version was latency-bound instead).  Row dot via
// a log2(DH/8)-step shuffle reduce within the row's thread group.
__global__ __launch_bounds__(256)
void attn_bwd_dot_kernel(const short* __restrict__ dO,
                         const short* __restrict__ O, float* __restrict__ Dl,
                         int H, int S, int DH, long n_rows) {
  const int tpr = DH >> 3;               // threads per row (8/16 for 64/128)
  const int rpb = 256 / tpr;             // rows per block
  const long row = (long)blockIdx.x * rpb + threadIdx.x / tpr;
  const int c8 = (threadIdx.x % tpr) * 8;
  float acc = 0.f;
  if (row < n_rows) {
    s16x8 dv = *(const s16x8*)(dO + row * DH + c8);
    s16x8 ov = *(const s16x8*)(O + row * DH + c8);
#pragma unroll
    for (int j = 0; j < 8; ++j) acc += bfbits2f(dv[j]) * bfbits2f(ov[j]);
  }
  // reduce across the row's thread group (contiguous lanes)
  for (int off = tpr >> 1; off > 0; off >>= 1)
    acc += __shfl_xor(acc, off);
  if (row < n_rows && (threadIdx.x % tpr) == 0) {
    // remap (b,s,h) -> (b,h,s) to match LSE layout
    long h = row % H;
    long s = (row / H) % S;
    long b = row / ((long)H * S);
    Dl[((b * H) + h) * S + s] = acc;
  }
}

// dK/dV kernel: one 4-wave block per 64-key tile; loops q-tiles of QT
// (32, or 64 at long sequence — halves the per-tile staging/sync fixed
// costs, which dominate the S=4096 backward).
template <int DH, int QT = KVT>
__global__ __launch_bounds__(256)
void attn_bwd_kv_kernel(const short* __restrict__ Q,
                        const short* __restrict__ K,
                        const short* __restrict__ V,
                        const short* __restrict__ dO,
                        const float* __restrict__ LSE,
                        const float* __restrict__ Dl,
                        const unsigned char* __restrict__ kv_pad,
                        short* __restrict__ dK, short* __restrict__ dV, int B,
                        int H, int Sq, int Sk, int causal, float scale,
                        long q_rs, long kv_rs, long do_rs, long dkv_rs) {
  constexpr int D32 = DH / 32;
  constexpr int D16 = DH / 16;
  constexpr int QH = QT / 16;          // 16-q halves per tile
  __shared__ short q_lds[QT * DH];     // q-tile natural [QT][DH]
  __shared__ short do_lds[QT * DH];    // dO-tile natural
  __shared__ short x_lds[WAVES][QW * QT];  // per-wave P^T / dS^T scratch

  const int bh = blockIdx.x;
  const int kb = blockIdx.y;           // key-block of 64
  const int b = bh / H, h = bh % H;
  const int wid = threadIdx.x >> 6, lane = threadIdx.x & 63;
  const int k0w = kb * (WAVES * QW) + wid * QW;  // wave's first key row
  const int fr = lane & 15, kg = lane >> 4;

  const short* Qp = Q + b * (Sq * q_rs) + (long)h * DH;
  const short* Kp = K + b * (Sk * kv_rs) + (long)h * DH;
  const short* Vp = V + b * (Sk * kv_rs) + (long)h * DH;
  const short* dOp = dO + b * (Sq * do_rs) + (long)h * DH;
  const float* lse = LSE + ((long)b * H + h) * Sq;
  const float* dl = Dl + ((long)b * H + h) * Sq;
  const unsigned char* pad = kv_pad ? kv_pad + (long)b * Sk : nullptr;

  // this wave's K,V rows in registers (A-fragments)
  bf16x8 kf[D32], vf[D32];
  const int krow = k0w + fr;
  const bool k_valid = krow < Sk;
  const bool k_pad = pad && k_valid && pad[krow];
#pragma unroll
  for (int d = 0; d < D32; ++d) {
    s16x8 kv_ = {0, 0, 0, 0, 0, 0, 0, 0}, vv = {0, 0, 0, 0, 0, 0, 0, 0};
    if (k_valid) {
      kv_ = *(const s16x8*)(Kp + (long)krow * kv_rs + d * 32 + kg * 8);
      vv = *(const s16x8*)(Vp + (long)krow * kv_rs + d * 32 + kg * 8);
    }
    kf[d] = (bf16x8)kv_;
    vf[d] = (bf16x8)vv;
  }

  f32x4 acc_dv[D16], acc_dk[D16];
#pragma unroll
  for (int i = 0; i < D16; ++i) {
    acc_dv[i] = {0, 0, 0, 0};
    acc_dk[i] = {0, 0, 0, 0};
  }

  // causal: q-rows before this key-block are fully masked
  const int q_start = causal ? (kb * (WAVES * QW)) / QT * QT : 0;

  // any pad among this wave's 16 OUTPUT keys (C rows k0w..k0w+15 — note
  // k_pad above is the lane's A-fragment row k0w+fr, a different role)
  bool wave_pad_any = false;
  if (pad) {
    const int kk = k0w + (lane & 15);
    wave_pad_any = __any(kk < Sk && pad[min(kk, Sk - 1)]);
  }

  for (int j0 = q_start; j0 < Sq; j0 += QT) {
    const int jc = min(QT, Sq - j0);
    stage_kv<DH, true, false, QT>(Qp + (long)j0 * q_rs, q_rs, jc,
                                  q_lds, nullptr);
    stage_kv<DH, true, false, QT>(dOp + (long)j0 * do_rs, do_rs, jc,
                                  do_lds, nullptr);
    __syncthreads();

    // pass 1: all halves of P^T into x_lds; dS^T kept in registers
    float ds_reg[QH][4];
#pragma unroll
    for (int half = 0; half < QH; ++half) {
      // S^T[key][q] = K·Q^T ; dP^T[key][q] = V·dO^T   (C row=key, col=q)
      f32x4 st = {0, 0, 0, 0}, dpt = {0, 0, 0, 0};
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int d = 0; d < D32; ++d) {
        bf16x8 qb_ = lds_read8<DH * 2>(q_lds, half * 16 + fr, d * 32 + kg * 8);
        bf16x8 dob = lds_read8<DH * 2>(do_lds, half * 16 + fr, d * 32 + kg * 8);
        st = __builtin_amdgcn_mfma_f32_16x16x32_bf16(kf[d], qb_, st, 0, 0, 0);
        dpt = __builtin_amdgcn_mfma_f32_16x16x32_bf16(vf[d], dob, dpt, 0, 0, 0);
      }
      __builtin_amdgcn_s_setprio(0);
      const int qcol = j0 + half * 16 + fr;
      const bool q_ok = qcol < Sq;
      const float lse_q = q_ok ? lse[min(qcol, Sq - 1)] : 0.f;
      const float d_q = q_ok ? dl[min(qcol, Sq - 1)] : 0.f;
      s16x4 pw;
      // per-lane clean: this lane's 4 OUTPUT keys (k0w+kg*4+r) valid,
      // no pad in the wave's keys, its q col in range, and (under
      // causal) all its keys visible to that col
      const bool clean2 = q_ok && !wave_pad_any &&
                          (k0w + kg * 4 + 3 < Sk) &&
                          (!causal || k0w + kg * 4 + 3 <= qcol);
      if (clean2) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          float p = __expf(st[r] * scale - lse_q);
          ds_reg[half][r] = p * (dpt[r] - d_q) * scale;
          pw[r] = f2bfbits(p);
        }
      } else {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int key = k0w + kg * 4 + r;
          float x = st[r] * scale;
          bool masked = !q_ok || key >= Sk || (pad && key < Sk && pad[key]) ||
                        (causal && key > qcol);
          float p = masked ? 0.f : __expf(x - lse_q);
          ds_reg[half][r] = p * (dpt[r] - d_q) * scale;
          pw[r] = f2bfbits(p);
        }
      }
      scrT_write4(x_lds[wid], half * 16 + fr, kg * 4, pw);
    }
    __builtin_amdgcn_wave_barrier();
    // dV += P^T · dO   (A = full P^T[key][q-chunk], B = dO^T[d][q])
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ks = 0; ks < QT / 32; ++ks) {
      bf16x8 pa = scrT_read8(x_lds[wid], ks * 32 + kg * 8);
#pragma unroll
      for (int i = 0; i < D16; ++i) {
        bf16x8 db = lds_read8_tr<DH * 2>(do_lds, ks * 32 + kg * 8, i * 16);
        acc_dv[i] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pa, db,
                                                            acc_dv[i],
                                                            0, 0, 0);
      }
    }
    __builtin_amdgcn_s_setprio(0);
    // pass 2: all halves of dS^T, then dK += dS^T · Q
#pragma unroll
    for (int half = 0; half < QH; ++half) {
      s16x4 dw = {f2bfbits(ds_reg[half][0]), f2bfbits(ds_reg[half][1]),
                  f2bfbits(ds_reg[half][2]), f2bfbits(ds_reg[half][3])};
      scrT_write4(x_lds[wid], half * 16 + fr, kg * 4, dw);
    }
    __builtin_amdgcn_wave_barrier();
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ks = 0; ks < QT / 32; ++ks) {
      bf16x8 sa = scrT_read8(x_lds[wid], ks * 32 + kg * 8);
#pragma unroll
      for (int i = 0; i < D16; ++i) {
        bf16x8 qb2 = lds_read8_tr<DH * 2>(q_lds, ks * 32 + kg * 8, i * 16);
        acc_dk[i] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(sa, qb2,
                                                            acc_dk[i],
                                                            0, 0, 0);
      }
    }
    __builtin_amdgcn_s_setprio(0);
    __syncthreads();
  }

  // write dK, dV (exclusive rows — no atomics)
  short* dKp = dK + b * (Sk * dkv_rs) + (long)h * DH;
  short* dVp = dV + b * (Sk * dkv_rs) + (long)h * DH;
#pragma unroll
  for (int i = 0; i < D16; ++i) {
    const int gcol = i * 16 + fr;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int key = k0w + kg * 4 + r;
      if (key >= Sk) continue;
      dKp[(long)key * dkv_rs + gcol] = f2bfbits(acc_dk[i][r]);
      dVp[(long)key * dkv_rs + gcol] = f2bfbits(acc_dv[i][r]);
    }
  }
}

// dQ kernel: one 4-wave block per 64-q tile; loops kv-tiles of QT.
template <int DH, int QT = KVT>
__global__ __launch_bounds__(256)
void attn_bwd_q_kernel(const short* __restrict__ Q,
                       const short* __restrict__ K,
                       const short* __restrict__ V,
                       const short* __restrict__ dO,
                       const float* __restrict__ LSE,
                       const float* __restrict__ Dl,
                       const unsigned char* __restrict__ kv_pad,
                       short* __restrict__ dQ, int B, int H, int Sq, int Sk,
                       int causal, float scale, long q_rs, long kv_rs,
                       long do_rs, long dq_rs) {
  constexpr int D32 = DH / 32;
  constexpr int D16 = DH / 16;
  constexpr int QH = QT / 16;
  __shared__ short k_lds[QT * DH];
  __shared__ short v_lds[QT * DH];
  __shared__ short x_lds[WAVES][QW * QT];

  const int bh = blockIdx.x;
  const int qb = blockIdx.y;
  const int b = bh / H, h = bh % H;
  const int wid = threadIdx.x >> 6, lane = threadIdx.x & 63;
  const int q0 = qb * (WAVES * QW) + wid * QW;
  const int fr = lane & 15, kg = lane >> 4;

  const short* Qp = Q + b * (Sq * q_rs) + (long)h * DH;
  const short* Kp = K + b * (Sk * kv_rs) + (long)h * DH;
  const short* Vp = V + b * (Sk * kv_rs) + (long)h * DH;
  const short* dOp = dO + b * (Sq * do_rs) + (long)h * DH;
  const float* lse = LSE + ((long)b * H + h) * Sq;
  const float* dl = Dl + ((long)b * H + h) * Sq;
  const unsigned char* pad = kv_pad ? kv_pad + (long)b * Sk : nullptr;

  bf16x8 qf[D32], dof[D32];
  const int qrow_l = q0 + fr;
  const bool q_valid = qrow_l < Sq;
#pragma unroll
  for (int d = 0; d < D32; ++d) {
    s16x8 qv = {0, 0, 0, 0, 0, 0, 0, 0}, dv = {0, 0, 0, 0, 0, 0, 0, 0};
    if (q_valid) {
      qv = *(const s16x8*)(Qp + (long)qrow_l * q_rs + d * 32 + kg * 8);
      dv = *(const s16x8*)(dOp + (long)qrow_l * do_rs + d * 32 + kg * 8);
    }
    qf[d] = (bf16x8)qv;
    dof[d] = (bf16x8)dv;
  }

  f32x4 acc_dq[D16];
#pragma unroll
  for (int i = 0; i < D16; ++i) acc_dq[i] = {0, 0, 0, 0};

  // this lane's 4 q-rows are FIXED for the whole kv loop — hoist their
  // lse/D values (they were re-loaded per tile x half: 2 scalar loads x
  // 4 rows x 2 halves x Sk/QT tiles)
  float lse_r[4], dl_r[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int qrow = q0 + kg * 4 + r;
    const bool ok = qrow < Sq;
    lse_r[r] = ok ? lse[min(qrow, Sq - 1)] : 0.f;
    dl_r[r] = ok ? dl[min(qrow, Sq - 1)] : 0.f;
  }

  const int kend = causal ? min(Sk, qb * (WAVES * QW) + WAVES * QW) : Sk;

  for (int k0 = 0; k0 < kend; k0 += QT) {
    const int kc = min(QT, Sk - k0);
    stage_kv<DH, true, false, QT>(Kp + (long)k0 * kv_rs, kv_rs, kc,
                                  k_lds, nullptr);
    stage_kv<DH, true, false, QT>(Vp + (long)k0 * kv_rs, kv_rs, kc,
                                  v_lds, nullptr);
    __syncthreads();

#pragma unroll
    for (int half = 0; half < QH; ++half) {
      f32x4 s = {0, 0, 0, 0}, dp = {0, 0, 0, 0};
      s16x4 dsw;
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int d = 0; d < D32; ++d) {
        bf16x8 kb_ = lds_read8<DH * 2>(k_lds, half * 16 + fr, d * 32 + kg * 8);
        bf16x8 vb_ = lds_read8<DH * 2>(v_lds, half * 16 + fr, d * 32 + kg * 8);
        s = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qf[d], kb_, s, 0, 0, 0);
        dp = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dof[d], vb_, dp, 0, 0, 0);
      }
      __builtin_amdgcn_s_setprio(0);
      const int kcol = k0 + half * 16 + fr;
      const bool col_pad = (kcol >= Sk) || (pad && pad[min(kcol, Sk - 1)]);
      // per-lane clean: col valid+unpadded, all 4 q rows in range and
      // (under causal) at/after this key col
      const bool clean2 = !col_pad && (q0 + kg * 4 + 3 < Sq) &&
                          (!causal || kcol <= q0 + kg * 4);
      if (clean2) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          float p = __expf(s[r] * scale - lse_r[r]);
          dsw[r] = f2bfbits(p * (dp[r] - dl_r[r]) * scale);
        }
      } else {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int qrow = q0 + kg * 4 + r;
          bool masked = (qrow >= Sq) || col_pad || (causal && kcol > qrow);
          float p = masked ? 0.f : __expf(s[r] * scale - lse_r[r]);
          dsw[r] = f2bfbits(p * (dp[r] - dl_r[r]) * scale);
        }
      }
      scrT_write4(x_lds[wid], half * 16 + fr, kg * 4, dsw);
    }
    __builtin_amdgcn_wave_barrier();
    // dQ += dS · K   (A = dS[q][key] from LDS, B = K^T[d][key])
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ks = 0; ks < QT / 32; ++ks) {
      bf16x8 sa = scrT_read8(x_lds[wid], ks * 32 + kg * 8);
#pragma unroll
      for (int i = 0; i < D16; ++i) {
        bf16x8 kb2 = lds_read8_tr<DH * 2>(k_lds, ks * 32 + kg * 8, i * 16);
        acc_dq[i] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(sa, kb2,
                                                            acc_dq[i],
                                                            0, 0, 0);
      }
    }
    __builtin_amdgcn_s_setprio(0);
    __syncthreads();
  }

  short* dQp = dQ + b * (Sq * dq_rs) + (long)h * DH;
#pragma unroll
  for (int i = 0; i < D16; ++i) {
    const int gcol = i * 16 + fr;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int qrow = q0 + kg * 4 + r;
      if (qrow >= Sq) continue;
      dQp[(long)qrow * dq_rs + gcol] = f2bfbits(acc_dq[i][r]);
    }
  }
}

// ---- host wrappers ---------------------------------------------------------

#define DISPATCH_DH(DH_VAL, ...)                                   \
  switch (DH_VAL) {                                                \
    case 32: { constexpr int DHC = 32; __VA_ARGS__; break; }       \
    case 64: { constexpr int DHC = 64; __VA_ARGS__; break; }       \
    case 128: { constexpr int DHC = 128; __VA_ARGS__; break; }     \
    default: TORCH_CHECK(false, "attention: head dim must be 32/64/128, got ", DH_VAL); \
  }

static void check_attn_view(const torch::Tensor& t, int DH, const char* name) {
  TORCH_CHECK(t.is_cuda() && t.dtype() == torch::kBFloat16 && t.dim() == 4,
              name, " must be 4-D bf16 on GPU");
  TORCH_CHECK(t.stride(3) == 1 && t.stride(2) == DH,
              name, " must have contiguous (head, dh) trailing layout");
}

// backward assumes batch stride == S * row stride (true for all training
// layouts: contiguous and packed-QKV/KV views); forward takes explicit batch
// strides so KV-cache slices (batch stride = S_max * row stride) work.
static void check_attn_batch(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.stride(0) == t.size(1) * t.stride(1),
              name, " batch stride must equal S * row stride");
}

// q (B,Sq,H,dh), k/v (B,Sk,H,dh) — possibly strided views (e.g. slots of a
// packed (B,S,3,H,dh) QKV tensor); k and v must share their row stride.
std::vector<torch::Tensor> attn_fwd(torch::Tensor q, torch::Tensor k,
                                    torch::Tensor v, torch::Tensor kv_pad,
                                    bool causal, double scale,
                                    int64_t trv) {
  const int B = q.size(0), Sq = q.size(1), H = q.size(2), DH = q.size(3);
  const int Sk = k.size(1);
  check_attn_view(q, DH, "q");
  check_attn_view(k, DH, "k");
  check_attn_view(v, DH, "v");
  TORCH_CHECK(k.stride(1) == v.stride(1), "k/v row strides must match");
  TORCH_CHECK(k.stride(0) == v.stride(0), "k/v batch strides must match");
  auto o = torch::empty({B, Sq, H, DH}, q.options());
  auto lse = torch::empty({B, H, Sq}, q.options().dtype(torch::kFloat32));
  const unsigned char* pad = nullptr;
  if (kv_pad.defined() && kv_pad.numel() > 0) {
    TORCH_CHECK(kv_pad.dtype() == torch::kUInt8 && kv_pad.is_contiguous() &&
                kv_pad.size(0) == B && kv_pad.size(1) == Sk);
    pad = kv_pad.data_ptr<unsigned char>();
  }
  // RF=2 (32 q-rows/wave) halves K/V re-staging but costs registers
  // (~188 vs ~130 VGPR -> 2 vs 3+ waves/SIMD).  MEASURED: end-to-end
  // seq-4096 training is ~6% SLOWER with it (L2 already absorbs much of
  // the re-read; occupancy is the binding constraint), so it stays
  // disabled; the RF template and its numerics test remain.
  const bool rf2 = false;
  dim3 grid(B * H, cdiv(Sq, WAVES * (rf2 ? 32 : 16)));
  auto stream = at::hip::getCurrentHIPStream();
  DISPATCH_DH(DH, {
    auto launch = [&](auto trv_c, auto rf_c) {
      attn_fwd_kernel<DHC, decltype(trv_c)::value, decltype(rf_c)::value>
          <<<grid, 256, 0, stream>>>(
              (const short*)q.data_ptr(), (const short*)k.data_ptr(),
              (const short*)v.data_ptr(), pad, (short*)o.data_ptr(),
              lse.data_ptr<float>(), B, H, Sq, Sk, causal ? 1 : 0,
              (float)scale, q.stride(1), k.stride(1), (long)H * DH,
              q.stride(0), k.stride(0));
    };
    using TT = std::true_type;
    using FF = std::false_type;
    using R1 = std::integral_constant<int, 1>;
    using R2 = std::integral_constant<int, 2>;
    if (trv) { if (rf2) launch(TT{}, R2{}); else launch(TT{}, R1{}); }
    else     { if (rf2) launch(FF{}, R2{}); else launch(FF{}, R1{}); }
  });
  return {o, lse};
}

// mode 0: dq, dk, dv separate contiguous tensors.
// mode 1: one packed dqkv (B,Sq,3,H,dh) — the self-attention path, gradient
//         flows straight into the packed-QKV linear with no cat/stack.
// mode 2: dq (B,Sq,H,dh) + packed dkv (B,Sk,2,H,dh) — the cross-attn path.
std::vector<torch::Tensor> attn_bwd(torch::Tensor q, torch::Tensor k,
                                    torch::Tensor v, torch::Tensor o,
                                    torch::Tensor dout, torch::Tensor lse,
                                    torch::Tensor kv_pad, bool causal,
                                    double scale, int64_t mode) {
  const int B = q.size(0), Sq = q.size(1), H = q.size(2), DH = q.size(3);
  const int Sk = k.size(1);
  check_attn_view(q, DH, "q");
  check_attn_view(k, DH, "k");
  check_attn_view(v, DH, "v");
  check_attn_batch(q, "q");
  check_attn_batch(k, "k");
  check_attn_batch(v, "v");
  TORCH_CHECK(k.stride(1) == v.stride(1), "k/v row strides must match");
  TORCH_CHECK(o.is_contiguous() && dout.is_contiguous(),
              "o/dout must be contiguous");
  std::vector<torch::Tensor> outs;
  short *dq_p, *dk_p, *dv_p;
  long dq_rs, dkv_rs;
  if (mode == 1) {
    TORCH_CHECK(Sq == Sk, "packed self mode requires Sq == Sk");
    auto dqkv = torch::empty({B, Sq, 3, H, DH}, q.options());
    short* base = (short*)dqkv.data_ptr();
    dq_p = base;
    dk_p = base + (long)H * DH;
    dv_p = base + 2L * H * DH;
    dq_rs = dkv_rs = 3L * H * DH;
    outs = {dqkv};
  } else if (mode == 2) {
    auto dq = torch::empty({B, Sq, H, DH}, q.options());
    auto dkv = torch::empty({B, Sk, 2, H, DH}, q.options());
    dq_p = (short*)dq.data_ptr();
    short* base = (short*)dkv.data_ptr();
    dk_p = base;
    dv_p = base + (long)H * DH;
    dq_rs = (long)H * DH;
    dkv_rs = 2L * H * DH;
    outs = {dq, dkv};
  } else {
    auto dq = torch::empty({B, Sq, H, DH}, q.options());
    auto dk = torch::empty({B, Sk, H, DH}, q.options());
    auto dv = torch::empty({B, Sk, H, DH}, q.options());
    dq_p = (short*)dq.data_ptr();
    dk_p = (short*)dk.data_ptr();
    dv_p = (short*)dv.data_ptr();
    dq_rs = dkv_rs = (long)H * DH;
    outs = {dq, dk, dv};
  }
  auto dl = torch::empty({B, H, Sq}, q.options().dtype(torch::kFloat32));
  const unsigned char* pad = nullptr;
  if (kv_pad.defined() && kv_pad.numel() > 0)
    pad = kv_pad.data_ptr<unsigned char>();
  auto stream = at::hip::getCurrentHIPStream();
  long n_rows = (long)B * Sq * H;
  attn_bwd_dot_kernel<<<cdiv(n_rows, 256 / (DH >> 3)), 256, 0, stream>>>(
      (const short*)dout.data_ptr(), (const short*)o.data_ptr(),
      dl.data_ptr<float>(), H, Sq, DH, n_rows);
  dim3 grid_kv(B * H, cdiv(Sk, WAVES * QW));
  dim3 grid_q(B * H, cdiv(Sq, WAVES * QW));
  const long q_rs = q.stride(1), kv_rs = k.stride(1);
  const long do_rs = (long)H * DH;
  // QT=64 at long sequence halves the per-tile staging/sync fixed costs
  // (the S=4096 backward's dominant overhead); QT=32 keeps short-seq
  // tail efficiency.  TFMX_ATTN_QT forces either for A/B.
  static const int qt_force = [] {
    const char* e = getenv("TFMX_ATTN_QT");
    return e ? atoi(e) : 0;
  }();
  const bool qt64 = qt_force ? (qt_force == 64)
                             : (Sq >= 1024 && Sk >= 1024);
  DISPATCH_DH(DH, {
    auto run = [&](auto qtc) {
      constexpr int QTC = decltype(qtc)::value;
      attn_bwd_kv_kernel<DHC, QTC><<<grid_kv, 256, 0, stream>>>(
          (const short*)q.data_ptr(), (const short*)k.data_ptr(),
          (const short*)v.data_ptr(), (const short*)dout.data_ptr(),
          lse.data_ptr<float>(), dl.data_ptr<float>(), pad, dk_p, dv_p, B,
          H, Sq, Sk, causal ? 1 : 0, (float)scale, q_rs, kv_rs, do_rs,
          dkv_rs);
      attn_bwd_q_kernel<DHC, QTC><<<grid_q, 256, 0, stream>>>(
          (const short*)q.data_ptr(), (const short*)k.data_ptr(),
          (const short*)v.data_ptr(), (const short*)dout.data_ptr(),
          lse.data_ptr<float>(), dl.data_ptr<float>(), pad, dq_p, B, H, Sq,
          Sk, causal ? 1 : 0, (float)scale, q_rs, kv_rs, do_rs, dq_rs);
    };
    if (qt64) run(std::integral_constant<int, 64>{});
    else run(std::integral_constant<int, 32>{});
  });
  return outs;
}
