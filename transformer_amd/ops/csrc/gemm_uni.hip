// Unified deep-pipelined 8-wave 256-row-tile bf16 GEMM for CDNA4 (gfx950).
//
// One schedule, two fragment modes per operand:
//   NT mode: operand stored [out][red] row-major (red contiguous) — staged
//            with the st_16x32 XOR swizzle, fragments via ds_read_b128
//            (gemm256.hip's layout).
//   TR mode: operand stored [red][out] row-major (out contiguous) — staged
//            RAW (coalesced) into [out-subtile][red-subtile][32 perm][16]
//            images, fragments via ds_read_b64_tr_b16 hardware transpose
//            reads (gemm_dw.hip's layout).
// Giving every training-shape GEMM a hand-written path with no transposes:
//   NT x NT : forward  C[M,N] = X[M,K] @ W[N,K]^T (+bias)(+ReLU)
//   NT x TR : dX       C[M,K] = dY[M,N] @ W[N,K]      (B read red-major)
//   TR x TR : dW       C[N,K] = dY[M,N]^T @ X[M,K]    (deep contraction)
// (reference contract rows K1/K7/K8/K12: Attention.py:46-50, point_ffn.py:
// 5-6, Transformer.py:16 — their backward GEMMs.)
//
// Schedule: 4 quadrant phases per K-tile (BK=64), glds prefetch of tile
// t+1 issued in the first two phases, one vmcnt(0) per K-tile at the
// boundary (the gemm256.hip schedule).  A full counted-vmcnt /
// consumption-ordered variant and per-quad / static s_setprio forms were
// built and MEASURED SLOWER on MI355X at every model shape
// (profiles/r2 sched A/B: counted 557-930 TF vs drain 600-1003; setprio
// -2..-4% — T5 is null-to-negative on this lockstep phase structure, and
// the per-tile drain already gives loads >=2.5 phases to land); they are
// kept as SCHED template variants for the bench tool only.
//
// Constraints (dispatched shapes all qualify; host falls back otherwise):
// K % 64 == 0, and TR-mode operands must be allocated out to the tile
// boundary (ragged-vocab tensors arrive 256-padded from ce_bwd /
// _wt_padded, so staging never reads past an allocation and no edge path
// exists); NT-mode out-row tails clamp, epilogue guards both tails.
#include "common.h"

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

typedef __bf16 bf16x8u __attribute__((ext_vector_type(8)));
typedef short s16x4u __attribute__((ext_vector_type(4)));

namespace {

#define U_BK 64
#define U_THREADS 512

// ---------------------------------------------------------------------------
// NT-mode layout (from gemm256.hip): [ROWS][64] bf16, 128-B rows, st_16x32
// swizzle phys = off ^ (((off>>9)&1)<<5); staged linear-dest glds with the
// inverse swizzle on the per-lane SOURCE address (guide rule 21).
// ---------------------------------------------------------------------------

DEV_INLINE int u_sw(int off) { return off ^ (((off >> 9) & 1) << 5); }

// Stage piece `p` of a ROWSx64 NT tile: chunks {h*(ROWS/16) + p*8 + wid}
// for h in 0..1 — rows [p*64+h*128 .. +64) at ROWS=256 (piece 0 = the rows
// quadrant phases q0 reads, piece 1 = q2's).  ROWS=128: single piece (p=0),
// chunks {h*8+wid}.  Rows clamp to maxrow-1 (duplicates discarded by the
// guarded epilogue) — NT out-rows only, never contraction.
template <int ROWS>
DEV_INLINE void u_stage_nt(const short* __restrict__ g, long ldg, int row0,
                           int maxrow, int p, short* lds) {
  const int wid = threadIdx.x >> 6, lane = threadIdx.x & 63;
#pragma unroll
  for (int h = 0; h < 2; ++h) {
    const int chunk = h * (ROWS / 16) + p * 8 + wid;
    if (ROWS == 128 && chunk >= 16) continue;
    const int d = chunk * 1024 + lane * 16;  // dest byte offset in tile
    const int lg = u_sw(d);                  // logical byte offset
    int row = row0 + (lg >> 7);              // 128-B rows
    if (row > maxrow - 1) row = maxrow - 1;
    const int colb = lg & 127;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)(g + (long)row * ldg +
                                                        (colb >> 1)),
        (__attribute__((address_space(3))) void*)((char*)lds + d), 16, 0, 0);
  }
}

DEV_INLINE bf16x8u u_frag_nt(const short* lds, int row, int kb) {
  const int off = u_sw(row * 128 + kb);
  return (bf16x8u)*(const s16x8*)((const char*)lds + off);
}

// ---------------------------------------------------------------------------
// TR-mode layout (from gemm_dw.hip): [out-subtile][red-subtile][32 perm][16]
// bf16 images; one [64 red][COLS out] tile = COLS/16 * 2 subtiles * 1 KiB.
// Within a subtile the red-row is stored permuted (pr = low3<<2 | top2) so
// the four 16-lane tr-read groups interleave across the 512-B subtile.
// ---------------------------------------------------------------------------

DEV_INLINE int u_img(int m, int col) {   // m = red row (0..63), col = out
  const int r = m & 31;
  const int pr = ((r & 7) << 2) | (r >> 3);
  return ((col >> 4) * 2 + (m >> 5)) * 512 + pr * 16 + (col & 15);
}

// glds staging (full blocks): dest lane-linear, inverse image permutation
// on the per-lane source address.  chunk c == subtile c = (cs)*2 + msub.
// piece p stages chunks {p*8 + half*16 + wid}; COLS=256 has pieces {0,1}
// (piece 0 = out cols the q0/q1 fragment reads touch first for A; for B all
// chunks are issued together).  COLS=128: single piece, chunks {half*8+wid}.
template <int COLS>
DEV_INLINE void u_stage_tr(const short* __restrict__ g, long ldg, long red0,
                           int out0, int p, short* lds) {
  const int wid = threadIdx.x >> 6, lane = threadIdx.x & 63;
#pragma unroll
  for (int half = 0; half < 2; ++half) {
    const int chunk = (COLS == 256) ? (p * 8 + half * 16 + wid)
                                    : (half * 8 + wid);
    const int d = chunk * 1024 + lane * 16;
    const int e = d >> 1;
    const int sub = e >> 9;
    const int we = e & 511;
    const int pr = we >> 4;
    const int r = ((pr & 3) << 3) | (pr >> 2);  // inverse permutation
    const int m = (sub & 1) * 32 + r;           // red row in tile
    const int col = (sub >> 1) * 16 + (we & 15);
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)(g + (red0 + m) * ldg +
                                                        out0 + col),
        (__attribute__((address_space(3))) void*)((char*)lds + d), 16, 0, 0);
  }
}

DEV_INLINE s16x4u u_tr4(const short* lds, int mb, int colb) {
  const int mp = threadIdx.x & 15;
  const int off = u_img(mb + (mp >> 2), colb + 4 * (mp & 3));
  return __builtin_amdgcn_ds_read_tr16_b64_v4i16(
      (__attribute__((address_space(3))) s16x4u*)(
          const_cast<short*>(&lds[off])));
}

DEV_INLINE bf16x8u u_frag_tr(const short* lds, int ms, int colb) {
  s16x4u lo = u_tr4(lds, ms, colb);
  s16x4u hi = u_tr4(lds, ms + 4, colb);
  s16x8 v = {lo[0], lo[1], lo[2], lo[3], hi[0], hi[1], hi[2], hi[3]};
  return (bf16x8u)v;
}

#define U_BARRIER() __builtin_amdgcn_s_barrier()
#define U_WAIT_LGKM0() asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory")
#define U_WAIT_VM(N) asm volatile("s_waitcnt vmcnt(" #N ")" ::: "memory")

// ---------------------------------------------------------------------------
// The kernel.  BM_ = 256 always (the 128 instance lives in gemm.hip);
// BN_ in {256, 128}.  TRA/TRB select fragment modes.  EPILOGUE: 0 none,
// 1 ReLU.  SPLITR: >0 = red axis sliced across blockIdx.y, fp32 partials.
// ---------------------------------------------------------------------------

// SCHED variants (A/B; tools/uni_bench.py — the default is the measured
// winner):
//   0 counted waits + per-quad s_setprio
//   1 counted waits, no setprio
//   2 r1-style: all A staged q0, B q1, vmcnt(0) at end-q3, no setprio
//   3 counted waits + static young-half priority (T5 static form)
//   4 counted waits, no setprio, B issued at q1 (balanced bursts)
template <int EPILOGUE, int BN_, bool TRA, bool TRB, bool SPLITR = false,
          int SCHED = 6>
__global__ __launch_bounds__(U_THREADS, 1)
void gemm_uni_kernel(const short* __restrict__ A, const short* __restrict__ B,
                     const short* __restrict__ bias, short* __restrict__ C,
                     float* __restrict__ CW,
                     int M, int N, int K, long lda, long ldb, int has_bias,
                     int nbm, int nbn, long k_per_slice) {
  constexpr int BM_ = 256;
  constexpr int MF = 8;                 // A 16-row frags per wave
  constexpr int NF = BN_ / 64;          // B frags per wave (4 or 2)
  constexpr int A_ELEMS = BM_ * U_BK;
  constexpr int B_ELEMS = BN_ * U_BK;
  constexpr int SLOT = A_ELEMS + B_ELEMS;
  constexpr int NB = BN_ / 64;          // B glds per thread per tile (4|2)
  extern __shared__ short smem[];
  const int lane = threadIdx.x & 63;

  // XCD-aware bijective workgroup remap (8 XCDs, private L2s).
  int nwg = nbm * nbn;
  int wg = blockIdx.x;
  {
    int q = nwg / 8, r = nwg % 8, x = wg % 8, o = wg / 8;
    wg = (x < r ? x * (q + 1) : r * (q + 1) + (x - r) * q) + o;
  }
  const int bm0 = (wg / nbn) * BM_;
  const int bn0 = (wg % nbn) * BN_;
  long k_lo = 0, k_hi = K;
  if (SPLITR) {
    k_lo = (long)blockIdx.y * k_per_slice;
    k_hi = min((long)K, k_lo + k_per_slice);
  }

  const int wid = threadIdx.x >> 6;
  const int wm = (wid >> 2) * (BM_ / 2);  // wave out-rows
  const int wn = (wid & 3) * (BN_ / 4);   // wave out-cols
  const int fr = lane & 15;
  const int kg = lane >> 4;               // red group 0..3

  f32x4 acc[MF][NF];
#pragma unroll
  for (int i = 0; i < MF; ++i)
#pragma unroll
    for (int j = 0; j < NF; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int ntiles = (int)((k_hi - k_lo) / U_BK);

  auto stage_a = [&](long kt, int piece, short* dst) {
    const long k0 = k_lo + kt * U_BK;
    if (TRA) {
      u_stage_tr<256>(A, lda, k0, bm0, piece, dst);
    } else {
      u_stage_nt<256>(A + k0, lda, bm0, M, piece, dst);
    }
  };
  auto stage_b = [&](long kt, short* dst) {
    const long k0 = k_lo + kt * U_BK;
    if (TRB) {
      if (BN_ == 256) {
        u_stage_tr<256>(B, ldb, k0, bn0, 0, dst);
        u_stage_tr<256>(B, ldb, k0, bn0, 1, dst);
      } else {
        u_stage_tr<128>(B, ldb, k0, bn0, 0, dst);
      }
    } else {
      if (BN_ == 256) {
        u_stage_nt<256>(B + k0, ldb, bn0, N, 0, dst);
        u_stage_nt<256>(B + k0, ldb, bn0, N, 1, dst);
      } else {
        u_stage_nt<128>(B + k0, ldb, bn0, N, 0, dst);
      }
    }
  };

  // Prologue: stage K-tile 0 into slot 0, drain once.
  stage_a(0, 0, smem);
  stage_a(0, 1, smem);
  stage_b(0, smem + A_ELEMS);
  U_WAIT_VM(0);
  U_BARRIER();

  if (SCHED == 3 && __builtin_amdgcn_readfirstlane(threadIdx.x) >= 256)
    __builtin_amdgcn_s_setprio(1);  // static young-half priority

  bf16x8u afr[MF / 2][2], bfr[NF][2];

  auto lda_frag = [&](int i, int ks, int rh, const short* a_lds) {
    const int row = wm + rh * (BM_ / 4) + i * 16;
    return TRA ? u_frag_tr(a_lds, ks * 32 + kg * 8, row)
               : u_frag_nt(a_lds, row + fr, ks * 64 + kg * 16);
  };
  auto ldb_frag = [&](int j, int ks, const short* b_lds) {
    const int col = wn + j * 16;
    return TRB ? u_frag_tr(b_lds, ks * 32 + kg * 8, col)
               : u_frag_nt(b_lds, col + fr, ks * 64 + kg * 16);
  };

#define U_MFMA_QUAD(RH, CH)                                                \
  if (SCHED == 0) __builtin_amdgcn_s_setprio(1);                          \
  _Pragma("unroll") for (int ks = 0; ks < 2; ++ks)                         \
    _Pragma("unroll") for (int i = 0; i < MF / 2; ++i)                     \
      _Pragma("unroll") for (int j = 0; j < NF / 2; ++j)                   \
        acc[(RH) * (MF / 2) + i][(CH) * (NF / 2) + j] =                    \
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(                       \
                afr[i][ks], bfr[(CH) * (NF / 2) + j][ks],                  \
                acc[(RH) * (MF / 2) + i][(CH) * (NF / 2) + j], 0, 0, 0);   \
  if (SCHED == 0) __builtin_amdgcn_s_setprio(0);

  for (int t = 0; t < ntiles; ++t) {
    const short* a_lds = smem + (t & 1) * SLOT;
    const short* b_lds = a_lds + A_ELEMS;
    short* pa_lds = smem + ((t + 1) & 1) * SLOT;
    short* pb_lds = pa_lds + A_ELEMS;
    const bool do_pf = t + 1 < ntiles;

    constexpr bool LOCKSTEP = (SCHED < 6);
    // ---- q0: quadrant (0,0); issue t+1's A-piece0 + all B -------------
#pragma unroll
    for (int i = 0; i < MF / 2; ++i)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) afr[i][ks] = lda_frag(i, ks, 0, a_lds);
#pragma unroll
    for (int j = 0; j < NF / 2; ++j)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) bfr[j][ks] = ldb_frag(j, ks, b_lds);
    if (do_pf) {
      stage_a(t + 1, 0, pa_lds);
      if (SCHED != 4) stage_b(t + 1, pb_lds);
      if (SCHED == 2 || SCHED >= 6) stage_a(t + 1, 1, pa_lds);
    }
    if (LOCKSTEP) U_BARRIER();
    U_WAIT_LGKM0();
    U_MFMA_QUAD(0, 0)
    if (LOCKSTEP) U_BARRIER();

    // ---- q1: quadrant (0,1); issue t+1's A-piece1 ---------------------
#pragma unroll
    for (int j = NF / 2; j < NF; ++j)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) bfr[j][ks] = ldb_frag(j, ks, b_lds);
    if (do_pf) {
      if (SCHED == 4) stage_b(t + 1, pb_lds);
      if (SCHED != 2 && SCHED < 6) stage_a(t + 1, 1, pa_lds);
    }
    if (LOCKSTEP) U_BARRIER();
    U_WAIT_LGKM0();
    U_MFMA_QUAD(0, 1)
    // end-q1 wait: drain THIS tile's A-piece1 (issued q1 of t-1, 4 phases
    // ago, consumed by q2's reads one barrier from here); t+1's
    // A-piece0+B (2+NB) + A-piece1 (2) stay in flight.
    if (SCHED != 2 && SCHED < 6) {
      if (do_pf) {
        if (BN_ == 256) U_WAIT_VM(8); else U_WAIT_VM(6);
      } else {
        U_WAIT_VM(0);
      }
    }
    if (LOCKSTEP || SCHED == 7) U_BARRIER();

    // ---- q2: quadrant (1,0) -------------------------------------------
#pragma unroll
    for (int i = 0; i < MF / 2; ++i)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) afr[i][ks] = lda_frag(i, ks, 1, a_lds);
    if (LOCKSTEP) U_BARRIER();
    U_WAIT_LGKM0();
    U_MFMA_QUAD(1, 0)
    if (LOCKSTEP) U_BARRIER();

    // ---- q3: quadrant (1,1), operands already in registers ------------
    U_MFMA_QUAD(1, 1)
    // end-q3 wait: drain t+1's A-piece0+B (issued q0, 3 phases ago);
    // its A-piece1 (2 glds) stays in flight across the tile boundary.
    if (SCHED == 2 || SCHED >= 6 || !do_pf) {
      U_WAIT_VM(0);
    } else {
      U_WAIT_VM(2);
    }
    U_BARRIER();
  }
#undef U_MFMA_QUAD

  // Epilogue: C/D lane map col = lane&15, row = (lane>>4)*4 + r.
#pragma unroll
  for (int i = 0; i < MF; ++i) {
    const int grow_base = bm0 + wm + i * 16 + kg * 4;
#pragma unroll
    for (int j = 0; j < NF; ++j) {
      const int gcol = bn0 + wn + j * 16 + fr;
      if (gcol >= N) continue;
      const float bv = (has_bias && bias) ? bfbits2f(bias[gcol]) : 0.0f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int grow = grow_base + r;
        if (grow >= M) continue;
        float v = acc[i][j][r] + bv;
        if (EPILOGUE == 1) v = fmaxf(v, 0.0f);
        if (SPLITR)
          CW[(long)blockIdx.y * M * N + (long)grow * N + gcol] = v;
        else
          C[(long)grow * N + gcol] = f2bfbits(v);
      }
    }
  }
}

__global__ void uni_reduce_kernel(const float* __restrict__ cw,
                                  short* __restrict__ c, long mn,
                                  int nslices) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= mn) return;
  float s = 0.f;
  for (int sl = 0; sl < nslices; ++sl) s += cw[(long)sl * mn + i];
  c[i] = f2bfbits(s);
}

}  // namespace

// Host-side viability: K deep enough and grid big enough for the pipeline.
bool gemm_uni_viable(int M, int N, int K) {
  if (K % U_BK != 0 || K < 4 * U_BK) return false;
  const int BN = ((long)cdiv(M, 256) * cdiv(N, 256) >= 224) ? 256 : 128;
  long nwg = (long)cdiv(M, 256) * cdiv(N, BN);
  return nwg >= 224 && (K >= 1024 || nwg >= 384);
}

// One launcher for all modes.  trA/trB: operand stored red-major.
// lda/ldb: leading dims in elements (>= the red extent for NT operands /
// the out extent for TR operands).
static torch::Tensor gemm_uni_launch(
    const torch::Tensor& a, const torch::Tensor& b,
    const c10::optional<torch::Tensor>& bias, int64_t epilogue,
    c10::optional<torch::Tensor> out, int M, int N, int K, long lda, long ldb,
    bool trA, bool trB, int splitr) {
  torch::Tensor c;
  if (out.has_value()) {
    c = *out;
    TORCH_CHECK(c.is_cuda() && c.dtype() == torch::kBFloat16 &&
                c.is_contiguous() && c.numel() == (long)M * N,
                "gemm_uni: bad out");
  } else {
    c = torch::empty({M, N}, a.options().dtype(torch::kBFloat16));
  }
  const bool has_bias = bias.has_value() && bias->defined() &&
                        bias->numel() > 0;
  TORCH_CHECK(K % U_BK == 0 && K >= 2 * U_BK, "gemm_uni: K must be n*64");
  TORCH_CHECK(splitr <= 1 || (epilogue == 0 && !has_bias),
              "gemm_uni: split-contraction excludes bias/activation");
  // widest tile whose grid still fills the 256-CU chip; prefer the
  // narrow tile when the wide grid is badly off a 256-block wave (the
  // 1-block/CU quantization cliff — see gemm256_tile)
  const long g22 = (long)cdiv(M, 256) * cdiv(N, 256);
  const long g21 = (long)cdiv(M, 256) * cdiv(N, 128);
  auto util = [](long nwg) {
    return nwg < 1 ? 0.0 : (double)nwg / ((nwg + 255) / 256 * 256);
  };
  const int BN = (g22 >= 224 && !(util(g21) > util(g22) + 0.1)) ? 256 : 128;
  const int nbm = cdiv(M, 256), nbn = cdiv(N, BN);
  // TR staging reads the full tile span of the out axis — the allocation
  // must cover it (ragged extents arrive padded: ce_bwd / _wt_padded).
  TORCH_CHECK(!trA || M % 256 == 0 || lda >= (long)cdiv(M, 256) * 256,
              "gemm_uni: TR A out extent not padded to the tile");
  TORCH_CHECK(!trB || N % BN == 0 || ldb >= (long)cdiv(N, BN) * BN,
              "gemm_uni: TR B out extent not padded to the tile");
  const size_t smem = 2 * ((size_t)256 + BN) * U_BK * sizeof(short);
  auto stream = at::hip::getCurrentHIPStream();

  int nslices = 1;
  long k_per_slice = K;
  torch::Tensor cw;
  float* cwp = nullptr;
  if (splitr > 1) {
    nslices = splitr;
    k_per_slice = ((K / U_BK + nslices - 1) / nslices) * U_BK;
    nslices = (int)((K + k_per_slice - 1) / k_per_slice);
    cw = torch::empty({(long)nslices * M * N},
                      a.options().dtype(torch::kFloat32));
    cwp = cw.data_ptr<float>();
  }
  dim3 grid(nbm * nbn, nslices);

  static bool attr_set[2][2][2][2][2] = {};
  auto launch = [&](auto epi, auto bnc, auto tra, auto trb, auto spl) {
    constexpr int E = decltype(epi)::value;
    constexpr int BNv = decltype(bnc)::value;
    constexpr bool TA = decltype(tra)::value;
    constexpr bool TB = decltype(trb)::value;
    constexpr bool SP = decltype(spl)::value;
    auto kfn = gemm_uni_kernel<E, BNv, TA, TB, SP>;
    bool& aset = attr_set[E][BNv == 128][TA][TB][SP];
    if (!aset) {
      (void)hipFuncSetAttribute((const void*)kfn,
                                hipFuncAttributeMaxDynamicSharedMemorySize,
                                (int)smem);
      aset = true;
    }
    kfn<<<grid, U_THREADS, smem, stream>>>(
        (const short*)a.data_ptr(), (const short*)b.data_ptr(),
        has_bias ? (const short*)bias->data_ptr() : nullptr,
        (short*)c.data_ptr(), cwp, M, N, K, lda, ldb, has_bias ? 1 : 0,
        nbm, nbn, k_per_slice);
  };
  using E0 = std::integral_constant<int, 0>;
  using E1 = std::integral_constant<int, 1>;
  using B256 = std::integral_constant<int, 256>;
  using B128 = std::integral_constant<int, 128>;
  using T = std::true_type;
  using F = std::false_type;
  // Only the mode combinations the framework dispatches are instantiated
  // (co-compiled variants perturb each other's codegen — guide rule 19):
  // NTxNT fwd (E0/E1), NTxTR dX (E0), TRxTR dW (E0, optionally split).
  auto d1 = [&](auto bnc) {
    if (trA && trB) {
      if (nslices > 1) launch(E0{}, bnc, T{}, T{}, T{});
      else launch(E0{}, bnc, T{}, T{}, F{});
    } else if (trB) {
      TORCH_CHECK(epilogue == 0, "gemm_uni: NTxTR carries no epilogue");
      launch(E0{}, bnc, F{}, T{}, F{});
    } else {
      TORCH_CHECK(!trA, "gemm_uni: TRxNT not instantiated");
      if (epilogue == 1) launch(E1{}, bnc, F{}, F{}, F{});
      else launch(E0{}, bnc, F{}, F{}, F{});
    }
  };
  if (BN == 256) d1(B256{}); else d1(B128{});
  if (nslices > 1) {
    long mn = (long)M * N;
    uni_reduce_kernel<<<(mn + 255) / 256, 256, 0, stream>>>(
        cwp, (short*)c.data_ptr(), mn, nslices);
  }
  return c;
}

// ---------------------------------------------------------------------------
// Ring-schedule NT GEMM (SCHED=5 in the A/B): ONE LDS slot, half-granular
// staging — each phase stages exactly one piece of tile t+1 into the
// region the previous phase's readers just vacated (piece death points:
// A-piece0 after q0's barrier, B after q1's, A-piece1 after q2's), with
// two counted s_waitcnt vmcnt(2) per K-tile (end-q1 covers A-piece1 of
// the CURRENT tile, end-q3 covers A-piece0+B of the NEXT) — the faithful
// reconstruction of the guide's 8-phase example's load pipeline.  Halves
// the LDS of the double-buffered schedule.
// ---------------------------------------------------------------------------

template <int EPILOGUE, int BN_, int WPS = 1>
__global__ __launch_bounds__(U_THREADS, WPS)
void gemm_ring_kernel(const short* __restrict__ A, const short* __restrict__ B,
                      const short* __restrict__ bias, short* __restrict__ C,
                      int M, int N, int K, long lda, long ldb, int has_bias,
                      int nbm, int nbn) {
  constexpr int BM_ = 256;
  constexpr int MF = 8;
  constexpr int NF = BN_ / 64;
  constexpr int A_ELEMS = BM_ * U_BK;
  extern __shared__ short smem[];
  const int lane = threadIdx.x & 63;

  int nwg = nbm * nbn;
  int wg = blockIdx.x;
  {
    int q = nwg / 8, r = nwg % 8, x = wg % 8, o = wg / 8;
    wg = (x < r ? x * (q + 1) : r * (q + 1) + (x - r) * q) + o;
  }
  const int bm0 = (wg / nbn) * BM_;
  const int bn0 = (wg % nbn) * BN_;

  const int wid = threadIdx.x >> 6;
  const int wm = (wid >> 2) * (BM_ / 2);
  const int wn = (wid & 3) * (BN_ / 4);
  const int fr = lane & 15;
  const int kg = lane >> 4;

  f32x4 acc[MF][NF];
#pragma unroll
  for (int i = 0; i < MF; ++i)
#pragma unroll
    for (int j = 0; j < NF; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int ntiles = K / U_BK;
  short* a_lds = smem;
  short* b_lds = smem + A_ELEMS;

  auto stage_a = [&](int kt, int piece) {
    u_stage_nt<256>(A + (long)kt * U_BK, lda, bm0, M, piece, a_lds);
  };
  auto stage_b = [&](int kt) {
    if (BN_ == 256) {
      u_stage_nt<256>(B + (long)kt * U_BK, ldb, bn0, N, 0, b_lds);
      u_stage_nt<256>(B + (long)kt * U_BK, ldb, bn0, N, 1, b_lds);
    } else {
      u_stage_nt<128>(B + (long)kt * U_BK, ldb, bn0, N, 0, b_lds);
    }
  };

  stage_a(0, 0);
  stage_a(0, 1);
  stage_b(0);
  U_WAIT_VM(0);
  U_BARRIER();

  bf16x8u afr[MF / 2][2], bfr[NF][2];

#define R_MFMA_QUAD(RH, CH)                                                \
  _Pragma("unroll") for (int ks = 0; ks < 2; ++ks)                         \
    _Pragma("unroll") for (int i = 0; i < MF / 2; ++i)                     \
      _Pragma("unroll") for (int j = 0; j < NF / 2; ++j)                   \
        acc[(RH) * (MF / 2) + i][(CH) * (NF / 2) + j] =                    \
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(                       \
                afr[i][ks], bfr[(CH) * (NF / 2) + j][ks],                  \
                acc[(RH) * (MF / 2) + i][(CH) * (NF / 2) + j], 0, 0, 0);

  for (int t = 0; t < ntiles; ++t) {
    const bool do_pf = t + 1 < ntiles;

    // q0: read A-piece0 + B-low fragments; MFMA (0,0)
#pragma unroll
    for (int i = 0; i < MF / 2; ++i)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
        afr[i][ks] = u_frag_nt(a_lds, wm + i * 16 + fr, ks * 64 + kg * 16);
#pragma unroll
    for (int j = 0; j < NF / 2; ++j)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
        bfr[j][ks] = u_frag_nt(b_lds, wn + j * 16 + fr, ks * 64 + kg * 16);
    U_BARRIER();
    U_WAIT_LGKM0();
    R_MFMA_QUAD(0, 0)
    U_BARRIER();
    // q1: stage A-piece0(t+1) into the region q0 vacated; read B-high
    if (do_pf) stage_a(t + 1, 0);
#pragma unroll
    for (int j = NF / 2; j < NF; ++j)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
        bfr[j][ks] = u_frag_nt(b_lds, wn + j * 16 + fr, ks * 64 + kg * 16);
    U_BARRIER();
    U_WAIT_LGKM0();
    R_MFMA_QUAD(0, 1)
    // end-q1: drain THIS tile's A-piece1 (staged q3 of t-1); t+1's
    // A-piece0 stays in flight
    if (do_pf) U_WAIT_VM(2); else U_WAIT_VM(0);
    U_BARRIER();
    // q2: stage B(t+1) over the dead B regions; read A-piece1
    if (do_pf) stage_b(t + 1);
#pragma unroll
    for (int i = 0; i < MF / 2; ++i)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
        afr[i][ks] = u_frag_nt(a_lds, wm + (BM_ / 4) + i * 16 + fr,
                               ks * 64 + kg * 16);
    U_BARRIER();
    U_WAIT_LGKM0();
    R_MFMA_QUAD(1, 0)
    U_BARRIER();
    // q3: stage A-piece1(t+1); MFMA (1,1) from registers
    if (do_pf) stage_a(t + 1, 1);
    R_MFMA_QUAD(1, 1)
    // end-q3: drain t+1's A-piece0+B (2-3 phases in flight); its
    // A-piece1 rides across the boundary (covered by next end-q1)
    if (do_pf) U_WAIT_VM(2); else U_WAIT_VM(0);
    U_BARRIER();
  }
#undef R_MFMA_QUAD

#pragma unroll
  for (int i = 0; i < MF; ++i) {
    const int grow_base = bm0 + wm + i * 16 + kg * 4;
#pragma unroll
    for (int j = 0; j < NF; ++j) {
      const int gcol = bn0 + wn + j * 16 + fr;
      if (gcol >= N) continue;
      const float bv = (has_bias && bias) ? bfbits2f(bias[gcol]) : 0.0f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int grow = grow_base + r;
        if (grow >= M) continue;
        float v = acc[i][j][r] + bv;
        if (EPILOGUE == 1) v = fmaxf(v, 0.0f);
        C[(long)grow * N + gcol] = f2bfbits(v);
      }
    }
  }
}

// Bench-only schedule A/B: NT x NT, no epilogue, BN=256 grid assumed.
torch::Tensor gemm_uni_nt_ab(torch::Tensor a, torch::Tensor w,
                             int64_t sched) {
  const int M = a.size(0), K = a.size(1), N = w.size(0);
  auto c = torch::empty({M, N}, a.options());
  const int BN = ((long)cdiv(M, 256) * cdiv(N, 256) >= 224) ? 256 : 128;
  const int nbm = cdiv(M, 256), nbn = cdiv(N, BN);
  const size_t smem = 2 * ((size_t)256 + BN) * U_BK * sizeof(short);
  auto stream = at::hip::getCurrentHIPStream();
  auto go = [&](auto bnc, auto sc) {
    constexpr int BNv = decltype(bnc)::value;
    constexpr int SC = decltype(sc)::value;
    auto kfn = gemm_uni_kernel<0, BNv, false, false, false, SC>;
    (void)hipFuncSetAttribute((const void*)kfn,
                              hipFuncAttributeMaxDynamicSharedMemorySize,
                              (int)smem);
    kfn<<<nbm * nbn, U_THREADS, smem, stream>>>(
        (const short*)a.data_ptr(), (const short*)w.data_ptr(), nullptr,
        (short*)c.data_ptr(), nullptr, M, N, K, K, K, 0, nbm, nbn, K);
  };
  auto go2 = [&](auto sc) {
    if (BN == 256) go(std::integral_constant<int, 256>{}, sc);
    else go(std::integral_constant<int, 128>{}, sc);
  };
  if ((int)sched == 5) {
    const size_t rsm = ((size_t)256 + BN) * U_BK * sizeof(short);
    auto ring = [&](auto bnc) {
      constexpr int BNv = decltype(bnc)::value;
      auto kfn = gemm_ring_kernel<0, BNv>;
      (void)hipFuncSetAttribute((const void*)kfn,
                                hipFuncAttributeMaxDynamicSharedMemorySize,
                                (int)rsm);
      kfn<<<nbm * nbn, U_THREADS, rsm, stream>>>(
          (const short*)a.data_ptr(), (const short*)w.data_ptr(), nullptr,
          (short*)c.data_ptr(), M, N, K, K, K, 0, nbm, nbn);
    };
    if (BN == 256) ring(std::integral_constant<int, 256>{});
    else ring(std::integral_constant<int, 128>{});
    return c;
  }
  if ((int)sched == 8) {
    // ring @ BN128 with a 4-waves/SIMD VGPR cap: 48 KiB LDS + <=128
    // VGPR admit TWO co-resident blocks per CU, overlapping prologue
    // drains and boundary stalls across blocks (the K=512 1-block/CU
    // shapes idle the CU during their prologue)
    const int nbn8 = cdiv(N, 128);
    const size_t rsm = ((size_t)256 + 128) * U_BK * sizeof(short);
    auto kfn = gemm_ring_kernel<0, 128, 4>;
    (void)hipFuncSetAttribute((const void*)kfn,
                              hipFuncAttributeMaxDynamicSharedMemorySize,
                              (int)rsm);
    kfn<<<nbm * nbn8, U_THREADS, rsm, stream>>>(
        (const short*)a.data_ptr(), (const short*)w.data_ptr(), nullptr,
        (short*)c.data_ptr(), M, N, K, K, K, 0, nbm, nbn8);
    return c;
  }
  switch ((int)sched) {
    case 0: go2(std::integral_constant<int, 0>{}); break;
    case 2: go2(std::integral_constant<int, 2>{}); break;
    case 3: go2(std::integral_constant<int, 3>{}); break;
    case 4: go2(std::integral_constant<int, 4>{}); break;
    case 6: go2(std::integral_constant<int, 6>{}); break;
    case 7: go2(std::integral_constant<int, 7>{}); break;
    default: go2(std::integral_constant<int, 1>{}); break;
  }
  return c;
}

// C[M,N] = A[M,K] @ B[N,K]^T (+bias)(+ReLU) — forward, NT x NT.
torch::Tensor gemm_uni_nt(torch::Tensor a, torch::Tensor w,
                          c10::optional<torch::Tensor> bias, int64_t epilogue,
                          c10::optional<torch::Tensor> out) {
  TORCH_CHECK(a.is_cuda() && a.dtype() == torch::kBFloat16 && a.dim() == 2 &&
              a.stride(1) == 1, "gemm_uni_nt: bad a");
  TORCH_CHECK(w.is_cuda() && w.dtype() == torch::kBFloat16 && w.dim() == 2 &&
              w.is_contiguous(), "gemm_uni_nt: bad w");
  const int M = a.size(0), K = a.size(1), N = w.size(0);
  TORCH_CHECK(w.size(1) == K, "gemm_uni_nt: K mismatch");
  return gemm_uni_launch(a, w, bias, epilogue, out, M, N, K,
                         a.stride(0), K, false, false, 1);
}

// C[M,N] = A[M,K] @ B[K,N] — dX = dY @ W, NT x TR (W read red-major via
// tr16, so the weight needs no transpose).  K = W's row count, %64 == 0.
torch::Tensor gemm_uni_nn(torch::Tensor a, torch::Tensor w,
                          c10::optional<torch::Tensor> out) {
  TORCH_CHECK(a.is_cuda() && a.dtype() == torch::kBFloat16 && a.dim() == 2 &&
              a.stride(1) == 1, "gemm_uni_nn: bad a");
  TORCH_CHECK(w.is_cuda() && w.dtype() == torch::kBFloat16 && w.dim() == 2 &&
              w.is_contiguous(), "gemm_uni_nn: bad w");
  const int M = a.size(0), K = a.size(1), N = w.size(1);
  TORCH_CHECK(w.size(0) == K, "gemm_uni_nn: K mismatch");
  return gemm_uni_launch(a, w, c10::nullopt, 0, out, M, N, K,
                         a.stride(0), N, false, true, 1);
}

// C[N,K] = A[M,N]^T @ B[M,K] — dW = dY^T @ X, TR x TR; contraction M%64==0.
// splitr > 1 slices the contraction across blockIdx.y with fp32 partials
// (for small-output deep-contraction shapes that would not fill the chip).
torch::Tensor gemm_uni_tn(torch::Tensor dy, torch::Tensor x,
                          c10::optional<torch::Tensor> out, int64_t splitr) {
  TORCH_CHECK(dy.is_cuda() && dy.dtype() == torch::kBFloat16 &&
              dy.dim() == 2 && dy.stride(1) == 1, "gemm_uni_tn: bad dy");
  TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kBFloat16 && x.dim() == 2 &&
              x.stride(1) == 1, "gemm_uni_tn: bad x");
  const long Mtok = dy.size(0);
  const int N = dy.size(1), K = x.size(1);
  TORCH_CHECK(x.size(0) == Mtok, "gemm_uni_tn: contraction mismatch");
  TORCH_CHECK(Mtok % U_BK == 0, "gemm_uni_tn: contraction must be n*64");
  return gemm_uni_launch(dy, x, c10::nullopt, 0, out, N, K, (int)Mtok,
                         dy.stride(0), x.stride(0), true, true,
                         (int)splitr);
}
