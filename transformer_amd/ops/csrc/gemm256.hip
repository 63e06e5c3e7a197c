// 256x256 8-wave deep-pipelined NT GEMM for CDNA4 (gfx950).
//
// The 128x128 two-barrier kernel in gemm.hip tops out near the "step-3
// structure" ceiling (~500-700 TF measured at model shapes): every
// __syncthreads() drains the global_load_lds queue (implicit vmcnt(0)), so
// staging latency is exposed once per K-step.  This kernel follows the
// guide's 8-phase 256^2 template instead: raw s_barrier + counted
// s_waitcnt vmcnt(N) (never 0 in the main loop) keep prefetch loads in
// flight across barriers.
//
// Geometry:
//   tile 256x256, BK=32, 512 threads = 8 waves in a 2(M) x 4(N) grid;
//   per-wave output 128x64 = acc[8][4] 16x16 fragments
//     (mfma_f32_16x16x32_bf16, fp32 accumulate).
//   LDS: ring of 4 K-tile slots x (A 16 KiB + B 16 KiB) = 128 KiB.
//   Per K-tile: 2 phases (one 64-row half of the wave's C each, 16 MFMAs);
//   each phase: 8-or-4 ds_read_b128 fragment loads, 2 global_load_lds
//   staging half-tiles of K-tile t+3 into slot (t+3)&3, s_barrier,
//   s_waitcnt lgkmcnt(0), s_setprio(1), 16 x MFMA, s_setprio(0), s_barrier;
//   at each tile boundary one s_waitcnt vmcnt(8) (loads from tiles t-1,t
//   may stay in flight; everything older - i.e. tile t+1's data - landed).
//
// LDS swizzle (st_16x32): phys = off ^ (((off>>9)&1)<<5) applied to the
// linear offset within each operand tile ([256 rows][32 k] bf16, 64-B
// rows).  16-lane ds_read_b128 column groups then hit 8 distinct 16-B
// slots per 256-B bank row (2-way conflict) instead of 4-way linear.
// Staged via global_load_lds with the inverse swizzle applied to the
// SOURCE address (LDS destination stays lane-linear).
//
// M/N edge handling: out-of-range tile rows are clamped to the last valid
// row at staging (duplicate data, discarded at the guarded epilogue), so
// any M,N work; K must be a multiple of 32 (all model shapes qualify
// except the V=32770 logits-dX contraction, which falls back to the
// 128x128 kernel).
#include "common.h"

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

typedef __bf16 bf16x8g __attribute__((ext_vector_type(8)));

namespace {

#define G_BM 256
#define G_BN 256
#define G_BK 32
#define G_THREADS 512
#define G_SLOTS 4
// shorts per operand tile (256 x 32)
#define G_TILE_ELEMS (G_BM * G_BK)
// shorts per slot (A tile + B tile)
#define G_SLOT_ELEMS (2 * G_TILE_ELEMS)

DEV_INLINE int sw256(int off) { return off ^ (((off >> 9) & 1) << 5); }

// Stage half h (tile rows [h*128, h*128+128)) of one 256x32 operand tile:
// 8 chunks of 1 KiB, one global_load_lds(16B) per thread.  `g` points at
// column k0 of the operand (row-major, ldg elems per row); rows are
// row0+tile_row clamped to maxrow-1.
DEV_INLINE void stage256(const short* __restrict__ g, long ldg, int row0,
                         int maxrow, int h, short* lds) {
  const int wid = threadIdx.x >> 6, lane = threadIdx.x & 63;
  const int chunk = h * 8 + wid;
  const int d = chunk * 1024 + lane * 16;  // dest byte offset in tile
  const int lg = sw256(d);                 // logical byte offset
  int row = row0 + (lg >> 6);              // 64-B rows
  if (row > maxrow - 1) row = maxrow - 1;
  const int colb = lg & 63;
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) void*)(g + (long)row * ldg +
                                                      (colb >> 1)),
      (__attribute__((address_space(3))) void*)((char*)lds + d), 16, 0, 0);
}

// Fragment load: A/B tile row `row` (0..255), k-group kg (0..3 -> 8 elems).
DEV_INLINE bf16x8g frag256(const short* lds, int row, int kg) {
  const int off = sw256(row * 64 + kg * 16);
  return (bf16x8g)*(const s16x8*)((const char*)lds + off);
}

#define G_BARRIER() __builtin_amdgcn_s_barrier()
#define G_WAIT_LGKM0() asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory")
#define G_WAIT_VM8() asm volatile("s_waitcnt vmcnt(8)" ::: "memory")
#define G_WAIT_VM0() asm volatile("s_waitcnt vmcnt(0)" ::: "memory")

template <int EPILOGUE>
__global__ __launch_bounds__(G_THREADS, 1)
void gemm256_kernel(const short* __restrict__ A, const short* __restrict__ B,
                    const short* __restrict__ bias, short* __restrict__ C,
                    int M, int N, int K, int lda, int ldb, int has_bias,
                    int nbm, int nbn) {
  extern __shared__ short smem[];
  const int lane = threadIdx.x & 63;

  // XCD-aware bijective workgroup remap (8 XCDs, private L2s).
  int nwg = nbm * nbn;
  int wg = blockIdx.x;
  {
    int q = nwg / 8, r = nwg % 8, x = wg % 8, o = wg / 8;
    wg = (x < r ? x * (q + 1) : r * (q + 1) + (x - r) * q) + o;
  }
  const int bm0 = (wg / nbn) * G_BM;
  const int bn0 = (wg % nbn) * G_BN;

  const int wid = threadIdx.x >> 6;
  const int wm = (wid >> 2) * 128;  // wave rows [wm, wm+128) of the C tile
  const int wn = (wid & 3) * 64;    // wave cols [wn, wn+64)
  const int fr = lane & 15;         // fragment lane row/col
  const int kg = lane >> 4;         // k-group 0..3

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int ntiles = K / G_BK;
  const short* Acol = A;  // advanced by k0 via explicit offset below
  const short* Bcol = B;

  // Prologue: stage K-tiles 0..2 into slots 0..2, full drain once.
#pragma unroll
  for (int t = 0; t < 3; ++t) {
    if (t < ntiles) {
      short* a_lds = smem + t * G_SLOT_ELEMS;
      short* b_lds = a_lds + G_TILE_ELEMS;
      const long k0 = (long)t * G_BK;
#pragma unroll
      for (int h = 0; h < 2; ++h) {
        stage256(Acol + k0, lda, bm0, M, h, a_lds);
        stage256(Bcol + k0, ldb, bn0, N, h, b_lds);
      }
    }
  }
  G_WAIT_VM0();
  G_BARRIER();

  bf16x8g af[4], bf_[4];
  for (int t = 0; t < ntiles; ++t) {
    const short* a_lds = smem + (t & 3) * G_SLOT_ELEMS;
    const short* b_lds = a_lds + G_TILE_ELEMS;
    const int pf = t + 3;          // prefetch tile
    short* pa_lds = smem + (pf & 3) * G_SLOT_ELEMS;
    short* pb_lds = pa_lds + G_TILE_ELEMS;
    const long pk0 = (long)pf * G_BK;
    const bool do_pf = pf < ntiles;

    // ---- phase 0: rows half rh=0 (frags 0..3), all 4 col frags --------
#pragma unroll
    for (int i = 0; i < 4; ++i)
      af[i] = frag256(a_lds, wm + i * 16 + fr, kg);
#pragma unroll
    for (int j = 0; j < 4; ++j)
      bf_[j] = frag256(b_lds, wn + j * 16 + fr, kg);
    if (do_pf) {
      stage256(Acol + pk0, lda, bm0, M, 0, pa_lds);
      stage256(Acol + pk0, lda, bm0, M, 1, pa_lds);
    }
    G_BARRIER();
    G_WAIT_LGKM0();
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j)
        acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            af[i], bf_[j], acc[i][j], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
    G_BARRIER();

    // ---- phase 1: rows half rh=1 (frags 4..7), B frags reused ---------
#pragma unroll
    for (int i = 0; i < 4; ++i)
      af[i] = frag256(a_lds, wm + 64 + i * 16 + fr, kg);
    if (do_pf) {
      stage256(Bcol + pk0, ldb, bn0, N, 0, pb_lds);
      stage256(Bcol + pk0, ldb, bn0, N, 1, pb_lds);
    }
    G_BARRIER();
    G_WAIT_LGKM0();
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j)
        acc[4 + i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            af[i], bf_[j], acc[4 + i][j], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
    // Tile boundary: everything staged before tiles t-1,t (8 loads/wave)
    // has landed -> tile t+1's slot is complete; never vmcnt(0).
    G_WAIT_VM8();
    G_BARRIER();
  }

  // Epilogue: C/D lane map col = lane&15, row = (lane>>4)*4 + r.
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    const int grow_base = bm0 + wm + i * 16 + kg * 4;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
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

}  // namespace

// Host-side eligibility check shared with the gemm.hip dispatcher.
bool gemm256_viable(int M, int N, int K, int lda, int ldb) {
  if (K % G_BK != 0 || K < 3 * G_BK) return false;
  if (lda % 8 != 0 || ldb % 8 != 0) return false;
  long nwg = (long)cdiv(M, G_BM) * cdiv(N, G_BN);
  // Measured rule (tools/gemm_bench.py on MI355X): the pipelined kernel
  // needs the chip full (>=224 WGs at 1 WG/CU) and either deep K or a
  // grid big enough to amortize the 3-tile prologue per WG.
  return nwg >= 224 && (K >= 1024 || nwg >= 512);
}

torch::Tensor gemm256_nt(torch::Tensor a, torch::Tensor w, torch::Tensor bias,
                         int64_t epilogue, c10::optional<torch::Tensor> out) {
  TORCH_CHECK(a.is_cuda() && a.dtype() == torch::kBFloat16 && a.dim() == 2 &&
              a.is_contiguous(), "gemm256_nt: a must be contiguous bf16 2-D");
  TORCH_CHECK(w.is_cuda() && w.dtype() == torch::kBFloat16 && w.dim() == 2 &&
              w.is_contiguous(), "gemm256_nt: w must be contiguous bf16 2-D");
  const int M = a.size(0), K = a.size(1), N = w.size(0);
  TORCH_CHECK(w.size(1) == K, "gemm256_nt: K mismatch");
  TORCH_CHECK(K % G_BK == 0 && K >= 3 * G_BK,
              "gemm256_nt: K must be a multiple of 32 (>=96)");
  const bool has_bias = bias.defined() && bias.numel() > 0;
  torch::Tensor c;
  if (out.has_value()) {
    c = *out;
    TORCH_CHECK(c.is_cuda() && c.dtype() == torch::kBFloat16 &&
                c.is_contiguous() && c.numel() == (long)M * N,
                "out must be contiguous bf16 with M*N elements");
  } else {
    c = torch::empty({M, N}, a.options());
  }
  const int nbm = cdiv(M, G_BM), nbn = cdiv(N, G_BN);
  const size_t smem = G_SLOTS * G_SLOT_ELEMS * sizeof(short);  // 128 KiB
  auto stream = at::hip::getCurrentHIPStream();
  static bool attr_set[2] = {false, false};
  auto launch = [&](auto epi) {
    constexpr int E = decltype(epi)::value;
    if (!attr_set[E]) {
      (void)hipFuncSetAttribute((const void*)gemm256_kernel<E>,
                                hipFuncAttributeMaxDynamicSharedMemorySize,
                                (int)smem);
      attr_set[E] = true;
    }
    gemm256_kernel<E><<<nbm * nbn, G_THREADS, smem, stream>>>(
        (const short*)a.data_ptr(), (const short*)w.data_ptr(),
        has_bias ? (const short*)bias.data_ptr() : nullptr,
        (short*)c.data_ptr(), M, N, K, K, K, has_bias, nbm, nbn);
  };
  if (epilogue == 1)
    launch(std::integral_constant<int, 1>{});
  else
    launch(std::integral_constant<int, 0>{});
  return c;
}
