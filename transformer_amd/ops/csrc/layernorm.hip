// Fused residual-add + LayerNorm fwd/bwd (semantics =
// reference Encoder.py:23 `layernorm(x + attn_output)` with eps=1e-6).
//
// One 64-lane wave per row (4 rows per 256-thread block); bf16 loads are
// vectorized short8 (guide G13), statistics and the backward reductions in
// fp32 via 64-wide shuffle reduces.  dgamma/dbeta accumulate into fp32
// workspaces with atomics, then cast to bf16.
#include "common.h"

#include <map>

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

DEV_INLINE unsigned int ln_pcg(unsigned long long key) {
  key = key * 6364136223846793005ull + 1442695040888963407ull;
  unsigned int h = (unsigned int)((key ^ (key >> 33)) >> 11);
  h ^= h >> 16; h *= 0x7feb352dU; h ^= h >> 15; h *= 0x846ca68bU;
  h ^= h >> 16;
  return h;
}

// forward: y = LN(drop(x)+res)*gamma+beta; saves s=drop(x)+res (bf16),
// mean & rstd (f32).  DROP fuses the sublayer dropout (reference
// Encoder.py:22 `dropout(attn)` feeding `layernorm(x + .)`) — one byte
// mask per element for the exact backward; DROP=false is plain
// residual+LN (inference / rate 0).
template <bool DROP>
__global__ __launch_bounds__(256)
void ln_fwd_kernel(const short* __restrict__ x, const short* __restrict__ res,
                   const short* __restrict__ gamma,
                   const short* __restrict__ beta, short* __restrict__ y,
                   short* __restrict__ s, float* __restrict__ mean_out,
                   float* __restrict__ rstd_out, int R, int D, float eps,
                   unsigned char* __restrict__ mask, float p, float inv_keep,
                   unsigned long long seed,
                   const long long* __restrict__ seed_ptr) {
  unsigned int thr = 0;
  if (DROP) {
    if (seed_ptr)
      seed = seed * 0xD1342543DE82EF95ull + (unsigned long long)(*seed_ptr);
    thr = (unsigned int)(p * 4294967296.0f);
  }
  const int row = blockIdx.x * 4 + (threadIdx.x >> 6);
  if (row >= R) return;
  const int lane = threadIdx.x & 63;
  const long base = (long)row * D;

  float sum = 0.f, sumsq = 0.f;
  // pass 1: s = drop(x) + res, accumulate stats
  for (int c = lane * 8; c < D; c += WAVE * 8) {
    if ((D % 8) == 0 && c + 8 <= D) {
      s16x8 xv = *(const s16x8*)(x + base + c);
      s16x8 rv = *(const s16x8*)(res + base + c);
      s16x8 sv;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float xj = bfbits2f(xv[j]);
        if (DROP) {
          unsigned char keep =
              ln_pcg(seed * 0x9E3779B97F4A7C15ull + (base + c + j)) >= thr;
          mask[base + c + j] = keep;
          xj = keep ? xj * inv_keep : 0.f;
        }
        float t = xj + bfbits2f(rv[j]);
        sv[j] = f2bfbits(t);
        sum += t;
        sumsq += t * t;
      }
      *(s16x8*)(s + base + c) = sv;
    } else {
      for (int j = 0; c + j < D; ++j) {
        float xj = bfbits2f(x[base + c + j]);
        if (DROP) {
          unsigned char keep =
              ln_pcg(seed * 0x9E3779B97F4A7C15ull + (base + c + j)) >= thr;
          mask[base + c + j] = keep;
          xj = keep ? xj * inv_keep : 0.f;
        }
        float t = xj + bfbits2f(res[base + c + j]);
        s[base + c + j] = f2bfbits(t);
        sum += t;
        sumsq += t * t;
      }
    }
  }
  sum = wave_sum(sum);
  sumsq = wave_sum(sumsq);
  const float mean = sum / D;
  const float var = fmaxf(sumsq / D - mean * mean, 0.f);
  const float rstd = rsqrtf(var + eps);
  if (lane == 0) {
    mean_out[row] = mean;
    rstd_out[row] = rstd;
  }
  // pass 2: normalize (s is in registers only partially; re-read via bf16)
  for (int c = lane * 8; c < D; c += WAVE * 8) {
    if ((D % 8) == 0 && c + 8 <= D) {
      s16x8 sv = *(const s16x8*)(s + base + c);
      s16x8 gv = *(const s16x8*)(gamma + c);
      s16x8 bv = *(const s16x8*)(beta + c);
      s16x8 yv;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        yv[j] = f2bfbits((bfbits2f(sv[j]) - mean) * rstd * bfbits2f(gv[j]) +
                         bfbits2f(bv[j]));
      *(s16x8*)(y + base + c) = yv;
    } else {
      for (int j = 0; c + j < D; ++j)
        y[base + c + j] =
            f2bfbits((bfbits2f(s[base + c + j]) - mean) * rstd *
                         bfbits2f(gamma[c + j]) + bfbits2f(beta[c + j]));
    }
  }
}

// backward: dx = rstd*(g - mean(g) - xhat*mean(g*xhat)), g = dy*gamma;
// dgamma += dy*xhat, dbeta += dy (atomics into fp32 workspace).
// DROP: additionally writes dxm = dx*mask*inv_keep — the gradient through
// the fused sublayer dropout (dx itself is the residual-branch gradient).
template <bool DROP>
__global__ __launch_bounds__(256)
void ln_bwd_kernel(const short* __restrict__ dy, const short* __restrict__ s,
                   const short* __restrict__ gamma,
                   const float* __restrict__ mean,
                   const float* __restrict__ rstd, short* __restrict__ dx,
                   int R, int D, const unsigned char* __restrict__ mask,
                   short* __restrict__ dxm, float inv_keep) {
  const int row = blockIdx.x * 4 + (threadIdx.x >> 6);
  if (row >= R) return;
  const int lane = threadIdx.x & 63;
  const long base = (long)row * D;
  const float mu = mean[row], rs = rstd[row];

  // s16x8 vector IO (the scalar per-lane version ran at ~1/3 of the
  // HBM bound); tail path for D % 8 != 0.
  float sg = 0.f, sgx = 0.f;
  for (int c = lane * 8; c < D; c += WAVE * 8) {
    if ((D % 8) == 0 && c + 8 <= D) {
      s16x8 dv = *(const s16x8*)(dy + base + c);
      s16x8 sv = *(const s16x8*)(s + base + c);
      s16x8 gv = *(const s16x8*)(gamma + c);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float xh = (bfbits2f(sv[j]) - mu) * rs;
        float g = bfbits2f(dv[j]) * bfbits2f(gv[j]);
        sg += g;
        sgx += g * xh;
      }
    } else {
      for (int j = 0; c + j < D; ++j) {
        float xh = (bfbits2f(s[base + c + j]) - mu) * rs;
        float g = bfbits2f(dy[base + c + j]) * bfbits2f(gamma[c + j]);
        sg += g;
        sgx += g * xh;
      }
    }
  }
  sg = wave_sum(sg) / D;
  sgx = wave_sum(sgx) / D;
  for (int c = lane * 8; c < D; c += WAVE * 8) {
    if ((D % 8) == 0 && c + 8 <= D) {
      s16x8 dv = *(const s16x8*)(dy + base + c);
      s16x8 sv = *(const s16x8*)(s + base + c);
      s16x8 gv = *(const s16x8*)(gamma + c);
      s16x8 xo, mo;
      unsigned long long mk = 0;
      if (DROP) mk = *(const unsigned long long*)(mask + base + c);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float xh = (bfbits2f(sv[j]) - mu) * rs;
        float g = bfbits2f(dv[j]) * bfbits2f(gv[j]);
        float d = rs * (g - sg - xh * sgx);
        xo[j] = f2bfbits(d);
        if (DROP)
          mo[j] = ((mk >> (8 * j)) & 0xff) ? f2bfbits(d * inv_keep)
                                           : (short)0;
      }
      *(s16x8*)(dx + base + c) = xo;
      if (DROP) *(s16x8*)(dxm + base + c) = mo;
    } else {
      for (int j = 0; c + j < D; ++j) {
        float xh = (bfbits2f(s[base + c + j]) - mu) * rs;
        float g = bfbits2f(dy[base + c + j]) * bfbits2f(gamma[c + j]);
        float d = rs * (g - sg - xh * sgx);
        dx[base + c + j] = f2bfbits(d);
        if (DROP)
          dxm[base + c + j] =
              mask[base + c + j] ? f2bfbits(d * inv_keep) : (short)0;
      }
    }
  }
}

// column reduction: dgamma[c] = sum_r dy[r][c]*xhat[r][c]; dbeta[c] = sum dy.
// Thread t of block b owns column b*256+t; every row read is coalesced
// across the block's 256 consecutive columns.
#define LNGB_ROWS 128
__global__ __launch_bounds__(256)
void ln_gb_kernel(const short* __restrict__ dy, const short* __restrict__ s,
                  const float* __restrict__ mean,
                  const float* __restrict__ rstd, float* __restrict__ acc_g,
                  float* __restrict__ acc_b, short* __restrict__ dgamma,
                  short* __restrict__ dbeta, int R, int D) {
  // 2 columns per thread (s16x2 loads) x 4-row unroll: the scalar
  // one-col-per-thread version was latency-bound at ~0.9 TB/s.
  const int c = (blockIdx.x * 64 + threadIdx.x) * 2;
  if (c < D) {
  const long r0 = (long)blockIdx.y * LNGB_ROWS;
  const long r1 = min((long)R, r0 + LNGB_ROWS);
  float sg0 = 0.f, sg1 = 0.f, sb0 = 0.f, sb1 = 0.f;
  const bool pair = (c + 1 < D) && (D % 2 == 0);
  if (pair) {
    long r = r0;
    for (; r + 4 <= r1; r += 4) {
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        s16x2 dv = *(const s16x2*)(dy + (r + u) * D + c);
        s16x2 sv = *(const s16x2*)(s + (r + u) * D + c);
        float mu = mean[r + u], rs = rstd[r + u];
        float d0 = bfbits2f(dv[0]), d1 = bfbits2f(dv[1]);
        sg0 += d0 * (bfbits2f(sv[0]) - mu) * rs;
        sg1 += d1 * (bfbits2f(sv[1]) - mu) * rs;
        sb0 += d0;
        sb1 += d1;
      }
    }
    for (; r < r1; ++r) {
      s16x2 dv = *(const s16x2*)(dy + r * D + c);
      s16x2 sv = *(const s16x2*)(s + r * D + c);
      float mu = mean[r], rs = rstd[r];
      float d0 = bfbits2f(dv[0]), d1 = bfbits2f(dv[1]);
      sg0 += d0 * (bfbits2f(sv[0]) - mu) * rs;
      sg1 += d1 * (bfbits2f(sv[1]) - mu) * rs;
      sb0 += d0;
      sb1 += d1;
    }
    atomicAdd(&acc_g[c], sg0);
    atomicAdd(&acc_b[c], sb0);
    atomicAdd(&acc_g[c + 1], sg1);
    atomicAdd(&acc_b[c + 1], sb1);
  } else {
    for (int cc = c; cc < min(c + 2, D); ++cc) {
      float sg = 0.f, sb = 0.f;
      for (long r = r0; r < r1; ++r) {
        float dyv = bfbits2f(dy[r * D + cc]);
        float xh = (bfbits2f(s[r * D + cc]) - mean[r]) * rstd[r];
        sg += dyv * xh;
        sb += dyv;
      }
      atomicAdd(&acc_g[cc], sg);
      atomicAdd(&acc_b[cc], sb);
    }
  }
  }
  // single-launch finalize: cast to bf16 + re-zero (was ln_gb_cast_kernel)
  if (last_arriver((unsigned*)(acc_b + D), gridDim.x * gridDim.y)) {
    for (int i = threadIdx.x; i < D; i += 64) {
      dgamma[i] = f2bfbits(acc_g[i]);
      dbeta[i] = f2bfbits(acc_b[i]);
      acc_g[i] = 0.f;
      acc_b[i] = 0.f;
    }
    if (threadIdx.x == 0)
      __hip_atomic_store((unsigned*)(acc_b + D), 0u, __ATOMIC_RELAXED,
                         __HIP_MEMORY_SCOPE_AGENT);
  }
}

// ---------------------------------------------------------------------------
// p > 0: dropout on x fused before the residual add; returns an extra
// byte mask tensor.  seed_t (optional int64[1] device counter) keeps
// HIP-graph replays stochastic, as in dropout_fwd.
std::vector<torch::Tensor> ln_fwd(torch::Tensor x, torch::Tensor res,
                                  torch::Tensor gamma, torch::Tensor beta,
                                  double eps, double p, int64_t seed,
                                  c10::optional<torch::Tensor> seed_t) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kBFloat16 && x.dim() == 2 &&
              x.is_contiguous() && res.is_contiguous());
  const int R = x.size(0), D = x.size(1);
  auto y = torch::empty_like(x);
  auto s = torch::empty_like(x);
  auto mean = torch::empty({R}, x.options().dtype(torch::kFloat32));
  auto rstd = torch::empty({R}, x.options().dtype(torch::kFloat32));
  auto stream = at::hip::getCurrentHIPStream();
  if (p > 0.0) {
    auto mask = torch::empty({(long)R * D}, x.options().dtype(torch::kUInt8));
    const long long* sp = nullptr;
    if (seed_t.has_value()) sp = (const long long*)seed_t->data_ptr();
    ln_fwd_kernel<true><<<cdiv(R, 4), 256, 0, stream>>>(
        (const short*)x.data_ptr(), (const short*)res.data_ptr(),
        (const short*)gamma.data_ptr(), (const short*)beta.data_ptr(),
        (short*)y.data_ptr(), (short*)s.data_ptr(), mean.data_ptr<float>(),
        rstd.data_ptr<float>(), R, D, (float)eps,
        mask.data_ptr<unsigned char>(), (float)p,
        1.0f / (1.0f - (float)p), (unsigned long long)seed, sp);
    return {y, s, mean, rstd, mask};
  }
  ln_fwd_kernel<false><<<cdiv(R, 4), 256, 0, stream>>>(
      (const short*)x.data_ptr(), (const short*)res.data_ptr(),
      (const short*)gamma.data_ptr(), (const short*)beta.data_ptr(),
      (short*)y.data_ptr(), (short*)s.data_ptr(), mean.data_ptr<float>(),
      rstd.data_ptr<float>(), R, D, (float)eps, nullptr, 0.f, 1.f, 0,
      nullptr);
  return {y, s, mean, rstd};
}

// mask given: additionally returns dxm = dx*mask*inv_keep (gradient of
// the fused sublayer dropout input); dx is the residual-branch gradient.
std::vector<torch::Tensor> ln_bwd(torch::Tensor dy, torch::Tensor s,
                                  torch::Tensor gamma, torch::Tensor mean,
                                  torch::Tensor rstd,
                                  c10::optional<torch::Tensor> dgamma_out,
                                  c10::optional<torch::Tensor> dbeta_out,
                                  c10::optional<torch::Tensor> mask,
                                  double p) {
  const int R = dy.size(0), D = dy.size(1);
  auto dx = torch::empty_like(dy);
  auto dest = [&](c10::optional<torch::Tensor>& o) {
    if (o.has_value()) {
      TORCH_CHECK(o->is_cuda() && o->dtype() == torch::kBFloat16 &&
                  o->is_contiguous() && o->numel() == D);
      return *o;
    }
    return torch::empty({D}, dy.options());
  };
  auto dgamma = dest(dgamma_out);
  auto dbeta = dest(dbeta_out);
  auto stream = at::hip::getCurrentHIPStream();
  torch::Tensor dxm;
  if (mask.has_value()) {
    dxm = torch::empty_like(dy);
    ln_bwd_kernel<true><<<cdiv(R, 4), 256, 0, stream>>>(
        (const short*)dy.data_ptr(), (const short*)s.data_ptr(),
        (const short*)gamma.data_ptr(), mean.data_ptr<float>(),
        rstd.data_ptr<float>(), (short*)dx.data_ptr(), R, D,
        mask->data_ptr<unsigned char>(), (short*)dxm.data_ptr(),
        1.0f / (1.0f - (float)p));
  } else {
    ln_bwd_kernel<false><<<cdiv(R, 4), 256, 0, stream>>>(
        (const short*)dy.data_ptr(), (const short*)s.data_ptr(),
        (const short*)gamma.data_ptr(), mean.data_ptr<float>(),
        rstd.data_ptr<float>(), (short*)dx.data_ptr(), R, D, nullptr,
        nullptr, 1.f);
  }
  static std::map<std::pair<int, int>, torch::Tensor> ws_cache;
  auto wkey = std::make_pair((int)dy.get_device(), D);
  auto wit = ws_cache.find(wkey);
  if (wit == ws_cache.end())
    wit = ws_cache.emplace(wkey, torch::zeros(
        {2L * D + 1}, dy.options().dtype(torch::kFloat32))).first;  // +cnt
  auto acc_g = wit->second.narrow(0, 0, D);
  auto acc_b = wit->second.narrow(0, D, D + 1);
  // 64-thread blocks: at d_model=512 a 256-thread block grid is only 128
  // workgroups — half the 256-CU chip idle.
  dim3 gbgrid(cdiv(cdiv(D, 2), 64), cdiv(R, LNGB_ROWS));
  ln_gb_kernel<<<gbgrid, 64, 0, stream>>>(
      (const short*)dy.data_ptr(), (const short*)s.data_ptr(),
      mean.data_ptr<float>(), rstd.data_ptr<float>(),
      acc_g.data_ptr<float>(), acc_b.data_ptr<float>(),
      (short*)dgamma.data_ptr(), (short*)dbeta.data_ptr(), R, D);
  if (mask.has_value()) return {dx, dgamma, dbeta, dxm};
  return {dx, dgamma, dbeta};
}
