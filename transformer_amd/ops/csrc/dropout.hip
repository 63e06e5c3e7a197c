// Dropout fwd/bwd (reference rate 0.1, Encoder.py:16-17).
// Counter-based RNG (PCG-style hash of (seed, index)) -> deterministic per
// (seed, element); mask saved as one byte per element for the exact
// backward.  y = x * mask / (1-p).
#include "common.h"

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

DEV_INLINE unsigned int pcg_hash(unsigned long long key) {
  key = key * 6364136223846793005ull + 1442695040888963407ull;
  unsigned int x = (unsigned int)((key ^ (key >> 33)) >> 11);
  x ^= x >> 16;
  x *= 0x7feb352dU;
  x ^= x >> 15;
  x *= 0x846ca68bU;
  x ^= x >> 16;
  return x;
}

// seed_ptr (optional): device step/epoch counter — under HIP-graph capture
// the host seed would be baked into the graph and every replay would
// reuse the same mask; reading the counter (advanced in-graph by
// adam_coefs_kernel) keeps replays stochastic.  `seed` doubles as a
// per-call-site salt on that path.
__global__ void dropout_fwd_kernel(const short* __restrict__ x,
                                   short* __restrict__ y,
                                   unsigned char* __restrict__ mask, long n,
                                   float p, float inv_keep,
                                   unsigned long long seed,
                                   const long long* __restrict__ seed_ptr) {
  long i = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 4;
  if (i >= n) return;
  if (seed_ptr)
    seed = seed * 0xD1342543DE82EF95ull + (unsigned long long)(*seed_ptr);
  const unsigned int thr = (unsigned int)(p * 4294967296.0f);
  if (i + 4 <= n) {                        // vector path (s16x4 + b32 mask)
    s16x4 xv = *(const s16x4*)(x + i);
    s16x4 yv;
    unsigned int mbits = 0;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      unsigned int r = pcg_hash(seed * 0x9E3779B97F4A7C15ull + (i + j));
      unsigned char keep = r >= thr;
      mbits |= (unsigned int)keep << (8 * j);
      yv[j] = keep ? f2bfbits(bfbits2f(xv[j]) * inv_keep) : (short)0;
    }
    *(s16x4*)(y + i) = yv;
    *(unsigned int*)(mask + i) = mbits;
    return;
  }
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    if (i + j >= n) break;
    unsigned int r = pcg_hash(seed * 0x9E3779B97F4A7C15ull + (i + j));
    unsigned char keep = r >= thr;
    mask[i + j] = keep;
    y[i + j] = keep ? f2bfbits(bfbits2f(x[i + j]) * inv_keep) : (short)0;
  }
}

__global__ void dropout_bwd_kernel(const short* __restrict__ dy,
                                   const unsigned char* __restrict__ mask,
                                   short* __restrict__ dx, long n,
                                   float inv_keep) {
  long i = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 4;
  if (i >= n) return;
  if (i + 4 <= n) {
    s16x4 dv = *(const s16x4*)(dy + i);
    unsigned int mbits = *(const unsigned int*)(mask + i);
    s16x4 xv;
#pragma unroll
    for (int j = 0; j < 4; ++j)
      xv[j] = ((mbits >> (8 * j)) & 0xff)
                  ? f2bfbits(bfbits2f(dv[j]) * inv_keep)
                  : (short)0;
    *(s16x4*)(dx + i) = xv;
    return;
  }
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    if (i + j >= n) break;
    dx[i + j] = mask[i + j] ? f2bfbits(bfbits2f(dy[i + j]) * inv_keep)
                            : (short)0;
  }
}

std::vector<torch::Tensor> dropout_fwd(torch::Tensor x, double p,
                                       int64_t seed,
                                       c10::optional<torch::Tensor> seed_t) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kBFloat16 &&
              x.is_contiguous());
  auto y = torch::empty_like(x);
  auto mask = torch::empty({x.numel()}, x.options().dtype(torch::kUInt8));
  long n = x.numel();
  const long long* sp = nullptr;
  if (seed_t.has_value()) {
    TORCH_CHECK(seed_t->is_cuda() && seed_t->dtype() == torch::kInt64 &&
                seed_t->numel() == 1);
    sp = (const long long*)seed_t->data_ptr();
  }
  auto stream = at::hip::getCurrentHIPStream();
  dropout_fwd_kernel<<<((n + 3) / 4 + 255) / 256, 256, 0, stream>>>(
      (const short*)x.data_ptr(), (short*)y.data_ptr(),
      mask.data_ptr<unsigned char>(), n, (float)p, 1.0f / (1.0f - (float)p),
      (unsigned long long)seed, sp);
  return {y, mask};
}

torch::Tensor dropout_bwd(torch::Tensor dy, torch::Tensor mask, double p) {
  auto dx = torch::empty_like(dy);
  long n = dy.numel();
  auto stream = at::hip::getCurrentHIPStream();
  dropout_bwd_kernel<<<((n + 3) / 4 + 255) / 256, 256, 0, stream>>>(
      (const short*)dy.data_ptr(), mask.data_ptr<unsigned char>(),
      (short*)dx.data_ptr(), n, 1.0f / (1.0f - (float)p));
  return dx;
}
