// 256x256 8-wave deep-pipelined NT GEMM for CDNA4 (gfx950) — the guide's
// "8-phase 256^2 template" reconstructed.
//
// Geometry: tile 256x256, BK=64, 512 threads = 8 waves in a 2(M) x 4(N)
// grid; per-wave output 128x64 = acc[8][4] 16x16 fragments
// (mfma_f32_16x16x32_bf16, fp32 accumulate).  LDS: TWO K-tile slots x
// (A 32 KiB + B 32 KiB) = 128 KiB.
//
// Per K-tile: 4 quadrant phases.  Each phase hoists its NEW operand
// fragments from LDS into registers (12/4/8/0 ds_read_b128 — A-halves and
// B-quadrant-halves are REUSED across phases from registers), prefetches
// half-tiles of K-tile t+1 into the other slot with global_load_lds, then
// s_waitcnt lgkmcnt(0) + 16 MFMAs.  NO intra-tile barriers (round 2,
// measured +5-6% per shape): within a tile every wave reads the SAME
// slot (synced at the boundary) and staging targets the other slot, so
// the per-phase barriers only enforced lockstep — dropping them lets
// waves drift and overlap each other's staging/MFMA phases (the 35%
// wave-parked time in profiles/r02_pmc_summary.md was these barriers).
// One vmcnt(0) + s_barrier per K-tile at the boundary remains:
//   q0: read A-half0(8) + B-ch0(4); stage A0',A1'  ; mfma quadrant (0,0)
//   q1: read B-ch1(4)             ; stage B0',B1'  ; mfma quadrant (0,1)
//   q2: read A-half1(8)           ;                ; mfma quadrant (1,0)
//   q3: (all in registers)        ; vmcnt(0)       ; mfma quadrant (1,1)
// The other slot is idle for the whole group (its tile was consumed last
// group), so all four prefetch halves are issued in the first two phases
// and have >=2.5 phases (~1400 cycles) to land — the boundary vmcnt(0)
// drains an (almost always) empty queue, unlike a __syncthreads() pipeline
// which exposes the full staging latency every K-step.
//
// LDS swizzle (st_16x32 on 128-B rows): phys = off ^ (((off>>9)&1)<<5) —
// 16-lane b128 column groups land on 4 distinct 16-B slots per bank row.
// Staged via global_load_lds with the inverse swizzle on the SOURCE address.
//
// M/N edges: out-of-range rows clamp to the last valid row at staging
// (duplicates, discarded by the guarded epilogue); K must be a multiple
// of 64 (all dispatched model shapes qualify; others use the 128 kernel).
#include "common.h"

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

typedef __bf16 bf16x8g __attribute__((ext_vector_type(8)));

namespace {

#define G_BM 256
#define G_BK 64
#define G_THREADS 512
// shorts per A tile (256 x 64); B tile is BN_ x 64 (BN_ = 256 or 128)
#define G_TILE_ELEMS (G_BM * G_BK)

DEV_INLINE int sw256(int off) { return off ^ (((off >> 9) & 1) << 5); }

// Stage half h (tile rows [h*ROWS/2, ...)) of one ROWSx64 operand tile:
// ROWS/8 chunks of 1 KiB, ROWS/128 global_load_lds(16B) per thread.
// `g` points at column k0 of the operand (row-major, ldg elems/row);
// rows clamped.
template <int ROWS = 256>
DEV_INLINE void stage256(const short* __restrict__ g, long ldg, int row0,
                         int maxrow, int h, short* lds) {
  const int wid = threadIdx.x >> 6, lane = threadIdx.x & 63;
#pragma unroll
  for (int p = 0; p < ROWS / 128; ++p) {
    const int chunk = h * (ROWS / 16) + p * 8 + wid;
    const int d = chunk * 1024 + lane * 16;  // dest byte offset in tile
    const int lg = sw256(d);                 // logical byte offset
    int row = row0 + (lg >> 7);              // 128-B rows
    if (row > maxrow - 1) row = maxrow - 1;
    const int colb = lg & 127;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)(g + (long)row * ldg +
                                                        (colb >> 1)),
        (__attribute__((address_space(3))) void*)((char*)lds + d), 16, 0, 0);
  }
}

// Fragment load: tile row `row` (0..255), k-group kg (0..3 -> 8 elems),
// k-step ks (0/1 -> elems 0..31 / 32..63).
DEV_INLINE bf16x8g frag256(const short* lds, int row, int kb /*bytes*/) {
  const int off = sw256(row * 128 + kb);
  return (bf16x8g)*(const s16x8*)((const char*)lds + off);
}

#define G_BARRIER() __builtin_amdgcn_s_barrier()
#define G_WAIT_LGKM0() asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory")
#define G_WAIT_VM0() asm volatile("s_waitcnt vmcnt(0)" ::: "memory")

// BM_/BN_ in {256, 128}: per-wave output (BM_/2) x (BN_/4); narrower
// instances keep the chip full on small-M/N shapes with the same
// pipelined phase structure.
template <int EPILOGUE, int BM_, int BN_, int ORDER = 0, int XCDMAP = 1>
__global__ __launch_bounds__(G_THREADS, 1)
void gemm256_kernel(const short* __restrict__ A, const short* __restrict__ B,
                    const short* __restrict__ bias, short* __restrict__ C,
                    int M, int N, int K, int lda, int ldb, int has_bias,
                    int nbm, int nbn) {
  constexpr int MF = BM_ / 32;          // A row-frags per wave (8 or 4)
  constexpr int NF = BN_ / 64;          // B col-frags per wave (4 or 2)
  constexpr int A_ELEMS = BM_ * G_BK;   // shorts per A tile
  constexpr int B_ELEMS = BN_ * G_BK;   // shorts per B tile
  constexpr int SLOT = A_ELEMS + B_ELEMS;
  extern __shared__ short smem[];
  const int lane = threadIdx.x & 63;

  // XCD-aware bijective workgroup remap (8 XCDs, private L2s) + tile
  // walk order (ORDER 0: m-major — consecutive wgs share the A panel;
  // 1: n-major — share the B panel.  At L3-resident huge-N shapes the
  // walk order decides which operand streams from HBM).
  int nwg = nbm * nbn;
  int wg = blockIdx.x;
  if (XCDMAP) {
    int q = nwg / 8, r = nwg % 8, x = wg % 8, o = wg / 8;
    wg = (x < r ? x * (q + 1) : r * (q + 1) + (x - r) * q) + o;
  }
  const int bm0 = (ORDER ? (wg % nbm) : (wg / nbn)) * BM_;
  const int bn0 = (ORDER ? (wg / nbm) : (wg % nbn)) * BN_;

  const int wid = threadIdx.x >> 6;
  const int wm = (wid >> 2) * (BM_ / 2);  // wave rows
  const int wn = (wid & 3) * (BN_ / 4);   // wave cols
  const int fr = lane & 15;         // fragment lane row/col
  const int kg = lane >> 4;         // k-group 0..3

  f32x4 acc[MF][NF];
#pragma unroll
  for (int i = 0; i < MF; ++i)
#pragma unroll
    for (int j = 0; j < NF; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int ntiles = K / G_BK;

  // Prologue: stage K-tile 0 into slot 0, drain once.
  {
    short* a0 = smem;
    short* b0 = smem + A_ELEMS;
#pragma unroll
    for (int h = 0; h < 2; ++h) {
      stage256<BM_>(A, lda, bm0, M, h, a0);
      stage256<BN_>(B, ldb, bn0, N, h, b0);
    }
  }
  G_WAIT_VM0();
  G_BARRIER();

  // Per-wave operand fragments in registers:
  //   afr[i][ks]: A rows (wm + rh*64 + i*16 + fr), one rh half at a time
  //   bfr[j][ks]: B rows (wn + j*16 + fr), all four col-frags live
  bf16x8g afr[MF / 2][2], bfr[NF][2];

  for (int t = 0; t < ntiles; ++t) {
    const short* a_lds = smem + (t & 1) * SLOT;
    const short* b_lds = a_lds + A_ELEMS;
    short* pa_lds = smem + ((t + 1) & 1) * SLOT;
    short* pb_lds = pa_lds + A_ELEMS;
    const long pk0 = (long)(t + 1) * G_BK;
    const bool do_pf = t + 1 < ntiles;

#define G_MFMA_QUAD(RH, CH)                                                \
  _Pragma("unroll") for (int ks = 0; ks < 2; ++ks)                         \
    _Pragma("unroll") for (int i = 0; i < MF / 2; ++i)                     \
      _Pragma("unroll") for (int j = 0; j < NF / 2; ++j)                   \
        acc[(RH) * (MF / 2) + i][(CH) * (NF / 2) + j] =                    \
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(                       \
                afr[i][ks], bfr[(CH) * (NF / 2) + j][ks],                  \
                acc[(RH) * (MF / 2) + i][(CH) * (NF / 2) + j], 0, 0, 0);

    // ---- q0: quadrant (rh=0, ch=0) ------------------------------------
#pragma unroll
    for (int i = 0; i < MF / 2; ++i)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
        afr[i][ks] = frag256(a_lds, wm + i * 16 + fr, ks * 64 + kg * 16);
#pragma unroll
    for (int j = 0; j < NF / 2; ++j)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
        bfr[j][ks] = frag256(b_lds, wn + j * 16 + fr, ks * 64 + kg * 16);
    if (do_pf) {
      stage256<BM_>(A + pk0, lda, bm0, M, 0, pa_lds);
      stage256<BM_>(A + pk0, lda, bm0, M, 1, pa_lds);
    }
    G_WAIT_LGKM0();
    G_MFMA_QUAD(0, 0)

    // ---- q1: quadrant (0, 1) ------------------------------------------
#pragma unroll
    for (int j = NF / 2; j < NF; ++j)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
        bfr[j][ks] = frag256(b_lds, wn + j * 16 + fr, ks * 64 + kg * 16);
    if (do_pf) {
      stage256<BN_>(B + pk0, ldb, bn0, N, 0, pb_lds);
      stage256<BN_>(B + pk0, ldb, bn0, N, 1, pb_lds);
    }
    G_WAIT_LGKM0();
    G_MFMA_QUAD(0, 1)

    // ---- q2: quadrant (1, 0) ------------------------------------------
#pragma unroll
    for (int i = 0; i < MF / 2; ++i)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
        afr[i][ks] = frag256(a_lds, wm + (BM_ / 4) + i * 16 + fr,
                             ks * 64 + kg * 16);
    G_WAIT_LGKM0();
    G_MFMA_QUAD(1, 0)

    // ---- q3: quadrant (1, 1), all operands already in registers -------
    G_MFMA_QUAD(1, 1)
    // Boundary: tile t+1's stages were issued >=2.5 phases ago; this
    // drain is (almost always) a no-op, unlike a per-K-step vmcnt(0).
    G_WAIT_VM0();
    G_BARRIER();
#undef G_MFMA_QUAD
  }

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
        C[(long)grow * N + gcol] = f2bfbits(v);
      }
    }
  }
}

}  // namespace

// Tile choice: widest instance whose grid still fills the 256-CU chip
// (narrower tiles re-read operands more, so wider wins when it fits —
// measured: BN=128 wins FFN2-class shapes +8% but loses on the logits dW
// where the wide grid already fills), EXCEPT when the wide grid lands
// badly off a multiple of 256 blocks: at 1 block/CU a 1.5-wave grid
// (e.g. QKV fwd: 64x6 = 384) idles a quarter of the chip for half the
// kernel — the narrower tile's full waves win despite the re-reads.
static void gemm256_tile(int M, int N, int& BM, int& BN) {
  const long g22 = (long)cdiv(M, 256) * cdiv(N, 256);
  const long g21 = (long)cdiv(M, 256) * cdiv(N, 128);
  auto util = [](long nwg) {
    return nwg < 1 ? 0.0 : (double)nwg / ((nwg + 255) / 256 * 256);
  };
  if (g22 >= 224) {
    BM = 256;
    BN = (util(g21) > util(g22) + 0.1) ? 128 : 256;
  }
  else if (g21 >= 224) { BM = 256; BN = 128; }
  else { BM = 128; BN = 128; }
}

// Host-side eligibility check shared with the gemm.hip dispatcher.
bool gemm256_viable(int M, int N, int K, int lda, int ldb) {
  if (K % G_BK != 0 || K < 2 * G_BK) return false;
  if (lda % 8 != 0 || ldb % 8 != 0) return false;
  int BM, BN;
  gemm256_tile(M, N, BM, BN);
  long nwg = (long)cdiv(M, BM) * cdiv(N, BN);
  // Measured rule (tools/gemm_bench.py, tools/probe128.py on MI355X):
  // needs the chip full and either moderate K depth or a grid big enough
  // to amortize the prologue.  (Round 2 relaxed K>=1024 to K>=512: with
  // the utilization-aware tile choice, gemm256 beats the 128-tile path
  // by 4-24% at the K=512 attn-O / cross-KV training shapes too.)
  return nwg >= 224 && (K >= 512 || nwg >= 384);
}

torch::Tensor gemm256_nt(torch::Tensor a, torch::Tensor w, torch::Tensor bias,
                         int64_t epilogue, c10::optional<torch::Tensor> out) {
  TORCH_CHECK(a.is_cuda() && a.dtype() == torch::kBFloat16 && a.dim() == 2 &&
              a.is_contiguous(), "gemm256_nt: a must be contiguous bf16 2-D");
  TORCH_CHECK(w.is_cuda() && w.dtype() == torch::kBFloat16 && w.dim() == 2 &&
              w.is_contiguous(), "gemm256_nt: w must be contiguous bf16 2-D");
  const int M = a.size(0), K = a.size(1), N = w.size(0);
  TORCH_CHECK(w.size(1) == K, "gemm256_nt: K mismatch");
  TORCH_CHECK(K % G_BK == 0 && K >= 2 * G_BK,
              "gemm256_nt: K must be a multiple of 64 (>=128)");
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
  int BMv, BNv;
  gemm256_tile(M, N, BMv, BNv);
  const int nbm = cdiv(M, BMv), nbn = cdiv(N, BNv);
  const size_t smem = 2 * ((size_t)BMv + BNv) * G_BK * sizeof(short);
  auto stream = at::hip::getCurrentHIPStream();
  static bool attr_set[2][3] = {};
  // tile-walk order / XCD-remap variants (TFMX_G256_ORDER=nm,
  // TFMX_G256_XCD=0 for the A/B prober; huge-N shapes pick n-major by
  // the measured rule below)
  static const int env_order = [] {
    const char* e = getenv("TFMX_G256_ORDER");
    return e ? (e[0] == 'n' ? 1 : 0) : -1;
  }();
  static const int env_xcd = [] {
    const char* e = getenv("TFMX_G256_XCD");
    return e ? atoi(e) : -1;
  }();
  const int order = env_order >= 0 ? env_order : 0;
  const int xcd = env_xcd >= 0 ? env_xcd : 1;
  auto launch = [&](auto epi, auto bmc, auto bnc) {
    constexpr int E = decltype(epi)::value;
    constexpr int BM_ = decltype(bmc)::value;
    constexpr int BN_ = decltype(bnc)::value;
    constexpr int ti = BM_ == 128 ? 2 : (BN_ == 128 ? 1 : 0);
    auto go = [&](auto oc, auto xc) {
      constexpr int OV = decltype(oc)::value;
      constexpr int XV = decltype(xc)::value;
      if (!attr_set[E][ti]) {
        (void)hipFuncSetAttribute(
            (const void*)gemm256_kernel<E, BM_, BN_, OV, XV>,
            hipFuncAttributeMaxDynamicSharedMemorySize, (int)smem);
        attr_set[E][ti] = true;
      }
      gemm256_kernel<E, BM_, BN_, OV, XV>
          <<<nbm * nbn, G_THREADS, smem, stream>>>(
              (const short*)a.data_ptr(), (const short*)w.data_ptr(),
              has_bias ? (const short*)bias.data_ptr() : nullptr,
              (short*)c.data_ptr(), M, N, K, K, K, has_bias, nbm, nbn);
    };
    using I0 = std::integral_constant<int, 0>;
    using I1 = std::integral_constant<int, 1>;
    if (order) { if (xcd) go(I1{}, I1{}); else go(I1{}, I0{}); }
    else       { if (xcd) go(I0{}, I1{}); else go(I0{}, I0{}); }
  };
  using E0 = std::integral_constant<int, 0>;
  using E1 = std::integral_constant<int, 1>;
  using T256 = std::integral_constant<int, 256>;
  using T128 = std::integral_constant<int, 128>;
  auto dis = [&](auto epi) {
    if (BMv == 128) launch(epi, T128{}, T128{});
    else if (BNv == 128) launch(epi, T256{}, T128{});
    else launch(epi, T256{}, T256{});
  };
  if (epilogue == 1) dis(E1{}); else dis(E0{});
  return c;
}
