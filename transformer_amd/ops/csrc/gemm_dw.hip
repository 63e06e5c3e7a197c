// Deep-contraction dW GEMM for CDNA4 (gfx950): C[N,K] = dY^T @ X.
//
// The linear-backward weight gradient contracts over tokens (M = 16384 for
// the base model) with small N,K (512..2048).  Both operands are stored
// token-major, so the MFMA fragments need contraction-contiguous (column)
// reads — the transpose problem.  Previous approaches measured on MI355X
// (tools/gemm_bench.py): physical transpose2d + NT kernel ~340 TF, LDS
// scratch-bounce TN ~230 TF, hipBLASLt TN ~240-330 TF at these shapes.
//
// This kernel instead stages dY/X blocks token-major (coalesced HBM reads)
// into [32 m][16 col] LDS subtiles and reads MFMA fragments directly with
// gfx950's ds_read_b64_tr_b16 hardware transpose-read (guide T10; lane
// semantics verified by tools/tr16_probe.hip):
//   per 16-lane group, the lane with local index m' supplies the address
//   &T[mb + m'/4][nb + 4*(m'%4)] and the lane with local index n receives
//   {T[mb+j][nb+n]}_{j=0..3} — a free 4x4 transpose, no scratch pass.
//
// Tiling: output 128(n) x 128(k) per 256-thread WG (4 waves, 64x64 each,
// acc[4][4] 16x16 frags); contraction in 64-token blocks (2 MFMA k-steps);
// the M axis is split across blockIdx.y slices writing DISJOINT fp32
// partials (an atomicAdd epilogue measured at the chip's fp32-atomic
// rate), reduced by a chip-filling pass, so the 256-CU chip stays full
// even when N=K=512 (16 output tiles).
#include "common.h"

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

typedef __bf16 bf16x8d __attribute__((ext_vector_type(8)));
typedef short s16x4d __attribute__((ext_vector_type(4)));

namespace {

#define DW_BN 128   // output rows (dY columns)
#define DW_BK 128   // output cols (X columns)
#define DW_BM 64    // contraction (token) block
#define DW_THREADS 256

// LDS image: [col-subtile][m-subtile][32 m][16 col] bf16; one operand block
// (64 m x 128 col) = 8 col-subtiles x 2 m-subtiles x 512 shorts = 16 KiB.
// Within a subtile the m-row is stored permuted (pr = low-3-bits<<2 | top-2)
// so the four 16-lane tr-read groups (m windows 8 apart) interleave across
// the 512-B subtile instead of stacking on the same 256-B bank row.
template <int ROWS = 64>
DEV_INLINE int dw_img(int m, int col) {
  const int r = m & 31;
  const int pr = ((r & 7) << 2) | (r >> 3);
  return ((col >> 4) * (ROWS / 32) + (m >> 5)) * 512 + pr * 16 + (col & 15);
}

// Stage a [64 m][128 col] block from global (row-major, ldg elems/row) into
// the subtile image.  Each of 256 threads: 4 x s16x8 loads (16 B, coalesced
// along col) + 4 ds_writes.  Out-of-range token rows and columns are
// ZERO-filled — rows are contraction terms here, so clamping (as the NT
// kernels do for discarded output rows) would add duplicate contributions.
template <int COLS = 128, int ROWS = 64>
DEV_INLINE void dw_stage(const short* __restrict__ g, long ldg, long m0,
                         long mmax, int col0, int ncols, short* lds) {
  const int t = threadIdx.x;
#pragma unroll
  for (int p = 0; p < ROWS * COLS / 2048; ++p) {
    int idx = p * DW_THREADS + t;      // ROWS*COLS/8 chunks of 8 shorts
    int m = idx / (COLS / 8);
    int c8 = (idx % (COLS / 8)) << 3;
    long gm = m0 + m;
    s16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
    if (gm < mmax) {
      if (col0 + c8 + 8 <= ncols) {
        v = *(const s16x8*)(g + gm * ldg + col0 + c8);
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j)
          if (col0 + c8 + j < ncols) v[j] = g[gm * ldg + col0 + c8 + j];
      }
    }
    // image dest: subtile row-half [m][c8..c8+8) — contiguous 16 B
    *(s16x4*)&lds[dw_img<ROWS>(m, c8)] = {v[0], v[1], v[2], v[3]};
    *(s16x4*)&lds[dw_img<ROWS>(m, c8 + 4)] = {v[4], v[5], v[6], v[7]};
  }
}

// Transpose-read one 4-deep fragment slice: returns T[mb..mb+4)[colb + n]
// for this lane's n-role (n = lane & 15).  mb must be a multiple of 4
// within one 32-m subtile.
template <int ROWS = 64>
DEV_INLINE s16x4d dw_tr4(const short* lds, int mb, int colb) {
  const int mp = threadIdx.x & 15;   // this lane's address role
  const int off = dw_img<ROWS>(mb + (mp >> 2), colb + 4 * (mp & 3));
  return __builtin_amdgcn_ds_read_tr16_b64_v4i16(
      (__attribute__((address_space(3))) s16x4d*)(
          const_cast<short*>(&lds[off])));
}

// Full 8-deep bf16x8 fragment: contraction ms..ms+8, column colb + (l&15).
template <int ROWS = 64>
DEV_INLINE bf16x8d dw_frag(const short* lds, int ms, int colb) {
  s16x4d lo = dw_tr4<ROWS>(lds, ms, colb);
  s16x4d hi = dw_tr4<ROWS>(lds, ms + 4, colb);
  s16x8 v = {lo[0], lo[1], lo[2], lo[3], hi[0], hi[1], hi[2], hi[3]};
  return (bf16x8d)v;
}

// glds staging for FULL blocks (64 valid m rows, 128 in-range cols): the
// image mapping is bijective, so the destination stays lane-linear and the
// inverse row permutation is applied to the per-lane SOURCE address — no
// register bounce, no ds_write pass (guide §5 rule 1).  Each wave stages 4
// of the 16 1-KiB chunks.
template <int COLS = 128, int ROWS = 64>
DEV_INLINE void dw_stage_glds(const short* __restrict__ g, long ldg, long m0,
                              int col0, short* lds) {
  const int wid = threadIdx.x >> 6, lane = threadIdx.x & 63;
#pragma unroll
  for (int p = 0; p < ROWS * COLS / 2048; ++p) {
    const int chunk = p * 4 + wid;
    const int d = chunk * 1024 + lane * 16;  // dest byte in image
    const int e = d >> 1;                    // dest element
    const int sub = e >> 9;
    const int we = e & 511;
    const int pr = we >> 4;
    const int r = ((pr & 3) << 3) | (pr >> 2);     // inverse permutation
    const int m = (sub % (ROWS / 32)) * 32 + r;
    const int col = (sub / (ROWS / 32)) * 16 + (we & 15);
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)(g + (m0 + m) * ldg +
                                                        col0 + col),
        (__attribute__((address_space(3))) void*)((char*)lds + d), 16, 0, 0);
  }
}

// DB: k-block-0 workgroups additionally accumulate the bias gradient
// db[n] = sum_m dY[m][n] into DBW (fp32, atomic) — dY is already streaming
// through this kernel, so the separate colsum pass (a full extra HBM read
// of dY per linear) disappears.  The re-read here hits L2: the same rows
// were just fetched by the glds staging.
// second launch_bounds arg pins >=4 waves/SIMD (VGPR cap 128): the DB
// variant otherwise allocates 134 VGPRs and drops a whole workgroup of
// block-level overlap per CU (measured -20% end-to-end).
typedef __bf16 bf16x4d __attribute__((ext_vector_type(4)));

// X16: use v_mfma_f32_16x16x16_bf16 (4-deep, 2-VGPR operands) so each
// ds_read_b64_tr_b16 result feeds an MFMA DIRECTLY — the 16x16x32 path
// pays ~65 v_mov per 64-m block pairing two v4i16 tr-reads into each
// 4-VGPR operand (seen in the compiled loop; ~24% of the MFMA issue
// time).  Same LDS traffic, twice the (half-size) MFMA instructions.
template <bool SPLIT, bool DB, bool NOBAR = true, bool X16 = false,
          int BM = DW_BM, int RING = 2>
__global__ __launch_bounds__(DW_THREADS)
void gemm_dw_kernel(const short* __restrict__ dY, const short* __restrict__ X,
                    short* __restrict__ C, float* __restrict__ CW,
                    float* __restrict__ DBW, int M,
                    int N, int K, long m_per_slice, int nbk) {
  // RING-slot staging ring (PMC: this kernel fetches each operand from
  // HBM once — L2 absorbs all tile re-reads — so it is pipeline-stall
  // bound, not traffic bound).  RING=2: prefetch block t+1, drain fully
  // at the boundary — but a 64-token block is only ~550 MFMA cycles,
  // under the ~900-cycle HBM latency, so the boundary drain stalls
  // (SQ_WAIT_ANY ~50%, profiles/r02_pmc_summary.md).  RING=3 stages
  // t+2 as well and waits the boundary with a COUNTED vmcnt (t+2's
  // loads ride across), trading LDS 64->96 KiB (2 -> 1 block/CU).
  __shared__ short a_img[RING][BM * DW_BN];  // dY block
  __shared__ short b_img[RING][BM * DW_BK];  // X block

  // XCD-aware bijective remap: consecutive logical tiles (same n-block,
  // varying k) land on the SAME XCD, so a dY slice is read into one XCD's
  // L2 once instead of once per k-tile (dY is the larger operand; the
  // blocks would otherwise round-robin across all 8 XCDs).
  int wg = blockIdx.x;
  {
    const int nwg = gridDim.x;
    const int q = nwg / 8, r = nwg % 8, x = wg % 8, o = wg / 8;
    wg = (x < r ? x * (q + 1) : r * (q + 1) + (x - r) * q) + o;
  }
  const int bn0 = (wg / nbk) * DW_BN;
  const int bk0 = (wg % nbk) * DW_BK;
  const long m_lo = (long)blockIdx.y * m_per_slice;
  const long m_hi = min((long)M, m_lo + m_per_slice);

  const int wid = threadIdx.x >> 6, lane = threadIdx.x & 63;
  const int wn = (wid >> 1) * 64;    // wave rows (n) in tile
  const int wk = (wid & 1) * 64;     // wave cols (k)
  const int fr = lane & 15;
  const int kg = lane >> 4;          // contraction group

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  // Full-block fast path: no m tail in range, both col blocks in range.
  const bool a_full = bn0 + DW_BN <= N;
  const bool b_full = bk0 + DW_BK <= K;

  // bias-grad accumulators: thread t covers 4 dY cols (bn0 + (t&31)*4),
  // 8 token rows per 64-m block — 4 fp32 regs so the DB variant stays
  // under the 4-workgroup VGPR budget (8 regs spilled or dropped a WG).
  const bool do_db = DB && bk0 == 0;
  const int db_c4 = (threadIdx.x & 31) << 2;
  float db_acc[4] = {0.f, 0.f, 0.f, 0.f};

  // staging helper: block starting at token row `mb` into slot `sl`
  auto stage_blk = [&](long mb, int sl) {
    if (mb >= m_hi) return;
    if (mb + BM <= m_hi && a_full)
      dw_stage_glds<128, BM>(dY, N, mb, bn0, a_img[sl]);
    else
      dw_stage<128, BM>(dY, N, mb, m_hi, bn0, N, a_img[sl]);
    if (mb + BM <= m_hi && b_full)
      dw_stage_glds<128, BM>(X, K, mb, bk0, b_img[sl]);
    else
      dw_stage<128, BM>(X, K, mb, m_hi, bk0, K, b_img[sl]);
  };

  // Prologue: stage block 0 (and, at RING=3, block 1), drain block 0.
  // The counted wait (vmcnt(8): block 1's 8 glds stay in flight) is only
  // valid when block 1 was glds-staged (full rows + full cols) — the
  // guarded register path issues no glds to count.
  stage_blk(m_lo, 0);
  if (RING == 3) stage_blk(m_lo + BM, 1 % RING);
  if (RING == 3 && a_full && b_full && m_lo + 2 * BM <= m_hi) {
    asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
  } else {
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  }
  __syncthreads();

  for (long m0 = m_lo; m0 < m_hi; m0 += BM) {
    if (do_db) {
#pragma unroll
      for (int p = 0; p < BM / 8; ++p) {
        const long gm = m0 + p * 8 + (threadIdx.x >> 5);
        if (gm < m_hi && bn0 + db_c4 + 4 <= N) {
          s16x4 v = *(const s16x4*)(dY + gm * N + bn0 + db_c4);
#pragma unroll
          for (int j = 0; j < 4; ++j) db_acc[j] += bfbits2f(v[j]);
        } else if (gm < m_hi) {
#pragma unroll
          for (int j = 0; j < 4; ++j)
            if (bn0 + db_c4 + j < N)
              db_acc[j] += bfbits2f(dY[gm * N + bn0 + db_c4 + j]);
        }
      }
    }
    const int cur = (int)((m0 - m_lo) / BM);
    const short* a_lds = a_img[cur % RING];
    const short* b_lds = b_img[cur % RING];
    // Prologue staged block 0; here prefetch block cur+1 into the other
    // slot (its previous tenant was consumed last block) — all glds
    // issued in phase 0, the rest of the block to land before the
    // boundary vmcnt(0) (template schedule, see gemm256.hip).
    const long mN = m0 + BM;
    const bool pf = mN < m_hi;
    bf16x8d af[4], bf_[4];
    s16x4d al[4], ah[4], bl[4], bh[4];
#pragma unroll
    for (int ph = 0; ph < BM / 32; ++ph) {
      const int ko = ph * 32 + kg * 8;
      if (!NOBAR && ph) __builtin_amdgcn_s_barrier();
      if (X16) {
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          al[i] = dw_tr4<BM>(a_lds, ko, wn + i * 16);
          ah[i] = dw_tr4<BM>(a_lds, ko + 4, wn + i * 16);
        }
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          bl[j] = dw_tr4<BM>(b_lds, ko, wk + j * 16);
          bh[j] = dw_tr4<BM>(b_lds, ko + 4, wk + j * 16);
        }
      } else {
#pragma unroll
        for (int i = 0; i < 4; ++i)
          af[i] = dw_frag<BM>(a_lds, ko, wn + i * 16);
#pragma unroll
        for (int j = 0; j < 4; ++j)
          bf_[j] = dw_frag<BM>(b_lds, ko, wk + j * 16);
      }
      if (ph == 0 && pf) {
        // stage the NEXT un-staged block: t+1 at RING=2, t+2 at RING=3
        const long mS = (RING == 3) ? mN + BM : mN;
        if (mS < m_hi) stage_blk(mS, (int)((cur + RING - 1) % RING));
      }
      // NOBAR: no intra-block barriers — all phases read the SAME LDS
      // slot (synced at the block boundary), fragments are wave-private
      // registers and the glds prefetch lands in the OTHER slot under
      // vmcnt; the compiler tracks the LDS-read -> MFMA dependencies and
      // software-pipelines the next phase's reads under this phase's
      // MFMA burst.  else: full s_barrier + lgkmcnt(0) drains per phase
      // keep the 4 waves in lockstep (A/B via TFMX_DW_BAR=1).
      if (!NOBAR) {
        __builtin_amdgcn_s_barrier();
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      }
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          if (X16) {
            acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x16bf16_1k(
                (bf16x4d)al[i], (bf16x4d)bl[j], acc[i][j], 0, 0, 0);
            acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x16bf16_1k(
                (bf16x4d)ah[i], (bf16x4d)bh[j], acc[i][j], 0, 0, 0);
          } else {
            acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                af[i], bf_[j], acc[i][j], 0, 0, 0);
          }
        }
    }
    // boundary: block t+1's stages must have landed.  RING=3 leaves
    // block t+2's 8 glds in flight across the barrier (vmcnt(8)) — valid
    // only when t+2 really was glds-staged (m0+3*BM fits and the tile is
    // full); otherwise drain (the guarded path's loads are
    // compiler-waited before its ds_writes, as at RING=2).
    if (RING == 3 && a_full && b_full && m0 + 3 * BM <= m_hi) {
      asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();
  }

  if (do_db) {
#pragma unroll
    for (int j = 0; j < 4; ++j)
      if (bn0 + db_c4 + j < N) atomicAdd(&DBW[bn0 + db_c4 + j], db_acc[j]);
  }

  // Epilogue: D lane map col = lane&15 -> k, row = (lane>>4)*4 + r -> n.
  // The fragment A row role is n (tr-read result lane = n), so output rows
  // follow the standard C/D mapping with rows = n, cols = k.
  // SPLIT epilogue writes DISJOINT per-slice fp32 partials (plain
  // stores): an atomicAdd version measured at the chip's fp32-atomic
  // rate (~8.4M atomics/call ~= the whole kernel time); the partials
  // are reduced by dw_reduce_kernel over a chip-filling N*K grid.
  float* cw_slice = SPLIT ? CW + (long)blockIdx.y * N * K : nullptr;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int nrow_base = bn0 + wn + i * 16 + kg * 4;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int kcol = bk0 + wk + j * 16 + fr;
      if (kcol >= K) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int nrow = nrow_base + r;
        if (nrow >= N) continue;
        if (SPLIT)
          cw_slice[(long)nrow * K + kcol] = acc[i][j][r];
        else
          C[(long)nrow * K + kcol] = f2bfbits(acc[i][j][r]);
      }
    }
  }
}

// Reduce the nslices disjoint fp32 partials -> bf16 C.  N*K is large
// (>=2^18 for every dispatched shape), so this grid fills the chip.
__global__ void dw_reduce_kernel(const float* __restrict__ cw,
                                 short* __restrict__ c, long nk,
                                 int nslices) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= nk) return;
  float s = 0.f;
  for (int sl = 0; sl < nslices; ++sl) s += cw[(long)sl * nk + i];
  c[i] = f2bfbits(s);
}

__global__ void dw_cast_kernel(const float* __restrict__ cw,
                               short* __restrict__ c, long nk) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < nk) c[i] = f2bfbits(cw[i]);
}

}  // namespace

// C[N,K] = dY[M,N]^T @ X[M,K], bf16 in/out, fp32 accumulate.  If `db` is
// given (bf16 [N]), the bias gradient sum_m dY[m][n] is produced in the
// same pass (no separate colsum read of dY).
torch::Tensor gemm_dw(torch::Tensor dy, torch::Tensor x,
                      c10::optional<torch::Tensor> out,
                      c10::optional<torch::Tensor> db,
                      c10::optional<torch::Tensor> cw_cached) {
  TORCH_CHECK(dy.is_cuda() && dy.dtype() == torch::kBFloat16 &&
              dy.dim() == 2 && dy.is_contiguous(), "gemm_dw: bad dy");
  TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kBFloat16 && x.dim() == 2 &&
              x.is_contiguous(), "gemm_dw: bad x");
  const long M = dy.size(0);
  const int N = dy.size(1), K = x.size(1);
  TORCH_CHECK(x.size(0) == M, "gemm_dw: contraction mismatch");
  torch::Tensor c;
  if (out.has_value()) {
    c = *out;
    TORCH_CHECK(c.is_cuda() && c.dtype() == torch::kBFloat16 &&
                c.is_contiguous() && c.numel() == (long)N * K,
                "gemm_dw: bad out");
  } else {
    c = torch::empty({N, K}, dy.options());
  }
  // Tile-size choice: the 256x256 8-wave kernel halves dY/X re-reads but
  // MEASURED SLOWER at every model shape (tools/gemm_bench.py: QKV dW
  // 0.155 vs 0.081 ms, FFN1 0.149 vs 0.076) — the extra split-M slices'
  // fp32 atomic traffic and the 8-wave barrier cost exceed the re-read
  // savings, so the 128-tile kernel is the only dispatch target.
  const int nbn = cdiv(N, DW_BN), nbk = cdiv(K, DW_BK);
  const int ntiles = nbn * nbk;
  // slices: fill >=DW_TARGET_WGS workgroups, 64-token quanta.  More slices
  // = more parallelism but more fp32 atomic traffic + per-WG prologue;
  // 512 measured best (sweepable via TFMX_DW_WGS for tools/gemm_bench.py).
  static int target_wgs = [] {
    const char* e = getenv("TFMX_DW_WGS");
    return e ? atoi(e) : 512;
  }();
  static const bool use_bm32 = [] {
    const char* e = getenv("TFMX_DW_BM");
    return e && atoi(e) == 32;
  }();
  const int bm = use_bm32 ? 32 : DW_BM;
  int nslices = (int)min((M + bm - 1) / bm,
                         (long)cdiv(target_wgs, ntiles));
  long m_per_slice = (M + nslices - 1) / nslices;
  m_per_slice = (m_per_slice + bm - 1) / bm * bm;
  nslices = (int)((M + m_per_slice - 1) / m_per_slice);
  dim3 grid(ntiles, nslices);
  auto stream = at::hip::getCurrentHIPStream();
  torch::Tensor dbw;
  float* dbw_p = nullptr;
  const bool has_db = db.has_value();
  if (has_db) {
    TORCH_CHECK(db->is_cuda() && db->dtype() == torch::kBFloat16 &&
                db->is_contiguous() && db->numel() == N, "gemm_dw: bad db");
    dbw = torch::zeros({N}, dy.options().dtype(torch::kFloat32));
    dbw_p = dbw.data_ptr<float>();
  }
  static const bool use_bar = [] {
    const char* e = getenv("TFMX_DW_BAR");
    return e && atoi(e) != 0;
  }();
  static const bool ring3 = [] {
    const char* e = getenv("TFMX_DW_RING");
    return e && atoi(e) == 3;
  }();
  static const bool use_x16 = [] {
    const char* e = getenv("TFMX_DW_X16");
    return e && atoi(e) != 0;
  }();
  auto launch = [&](auto split, auto dbc, float* cwp) {
    constexpr bool SP = decltype(split)::value, DBV = decltype(dbc)::value;
    auto kfn = use_bm32
        ? gemm_dw_kernel<SP, DBV, true, false, 32>
        : (ring3
               ? gemm_dw_kernel<SP, DBV, true, false, DW_BM, 3>
               : (use_x16
                      ? (use_bar ? gemm_dw_kernel<SP, DBV, false, true>
                                 : gemm_dw_kernel<SP, DBV, true, true>)
                      : (use_bar ? gemm_dw_kernel<SP, DBV, false, false>
                                 : gemm_dw_kernel<SP, DBV, true, false>)));
    kfn<<<grid, DW_THREADS, 0, stream>>>(
            (const short*)dy.data_ptr(), (const short*)x.data_ptr(),
            (short*)c.data_ptr(), cwp, dbw_p, (int)M, N, K, m_per_slice,
            nbk);
  };
  using T = std::true_type;
  using F = std::false_type;
  if (nslices == 1) {
    if (has_db) launch(F{}, T{}, nullptr); else launch(F{}, F{}, nullptr);
  } else {
    // Disjoint per-slice fp32 partials, plain stores, no zeroing needed
    // (every element of every slice is written); reduced by a
    // chip-filling pass.  `cw_cached` (an optional caller workspace from
    // the old atomic scheme) is reused as backing storage when large
    // enough.
    long nk = (long)N * K;
    torch::Tensor cw;
    if (cw_cached.has_value() &&
        cw_cached->numel() >= (long)nslices * nk &&
        cw_cached->dtype() == torch::kFloat32) {
      cw = *cw_cached;
    } else {
      cw = torch::empty({(long)nslices * nk},
                        dy.options().dtype(torch::kFloat32));
    }
    if (has_db) launch(T{}, T{}, cw.data_ptr<float>());
    else launch(T{}, F{}, cw.data_ptr<float>());
    dw_reduce_kernel<<<(nk + 255) / 256, 256, 0, stream>>>(
        cw.data_ptr<float>(), (short*)c.data_ptr(), nk, nslices);
  }
  if (has_db)
    dw_cast_kernel<<<cdiv(N, 256), 256, 0, stream>>>(
        dbw.data_ptr<float>(), (short*)db->data_ptr(), N);
  return c;
}
