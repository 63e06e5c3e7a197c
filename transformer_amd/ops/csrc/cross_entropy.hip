// Fused padding-masked (label-smoothed) cross entropy over the V≈32k vocab
// (semantics = reference train.py:83-88 with SURVEY §8 Q9
// smoothing: eps=0 reproduces plain CE, loss = sum(per-token)/global_batch).
//
// One 256-thread block (4 waves) per logits row; bf16 logits are read with
// short8 vector loads, max/sum reduced wave-wise then across the block's 4
// waves through LDS.  Forward saves the fp32 LSE per row; backward
// recomputes softmax from logits+LSE (no S×V weight materialization).
#include "common.h"

#include <map>

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

// block-level reduce over 4 waves (256 threads)
DEV_INLINE float block_reduce(float v, float* scratch, int op /*0=max 1=sum*/) {
  int wid = threadIdx.x >> 6, lane = threadIdx.x & 63;
  v = (op == 0) ? wave_max(v) : wave_sum(v);
  if (lane == 0) scratch[wid] = v;
  __syncthreads();
  float r = scratch[0];
#pragma unroll
  for (int i = 1; i < 4; ++i)
    r = (op == 0) ? fmaxf(r, scratch[i]) : r + scratch[i];
  __syncthreads();
  return r;
}

__global__ __launch_bounds__(256)
void ce_fwd_kernel(const short* __restrict__ logits,
                   const long* __restrict__ targets,
                   float* __restrict__ lse_out,
                   float* __restrict__ loss_banks,
                   long R, int V, float inv_batch, float eps_ls) {
  __shared__ float scratch[4];
  const long row = blockIdx.x;
  if (row >= R) return;
  const short* lrow = logits + row * V;
  const int t = threadIdx.x;

  // ONE pass over the 32k-vocab row (the two-pass version reads the
  // whole row twice — the second read is partly L2-resident, but this
  // kernel is still logits-read bound).  Branchless per-chunk online
  // rescale: always `s = s*exp(m_old - m_new) + chunk_sum` (exp of <=0),
  // then an (m, s) pair merge across lanes and waves.
  float mx = -1e30f, sume = 0.f, sumx = 0.f;
  for (int c = t * 8; c + 8 <= V; c += 256 * 8) {
    s16x8 v = *(const s16x8*)(lrow + c);
    float x0 = bfbits2f(v[0]), x1 = bfbits2f(v[1]);
    float x2 = bfbits2f(v[2]), x3 = bfbits2f(v[3]);
    float x4 = bfbits2f(v[4]), x5 = bfbits2f(v[5]);
    float x6 = bfbits2f(v[6]), x7 = bfbits2f(v[7]);
    float cm = fmaxf(fmaxf(fmaxf(x0, x1), fmaxf(x2, x3)),
                     fmaxf(fmaxf(x4, x5), fmaxf(x6, x7)));
    float mn = fmaxf(mx, cm);
    float cs = __expf(x0 - mn) + __expf(x1 - mn) + __expf(x2 - mn) +
               __expf(x3 - mn) + __expf(x4 - mn) + __expf(x5 - mn) +
               __expf(x6 - mn) + __expf(x7 - mn);
    sume = sume * __expf(mx - mn) + cs;
    mx = mn;
    sumx += ((x0 + x1) + (x2 + x3)) + ((x4 + x5) + (x6 + x7));
  }
  // V tail (V % 2048 partial chunk for this thread)
  {
    int c = ((V / 2048) * 2048) + t * 8;  // last sweep start for thread t
    if (c < V && c + 8 > V) {
      for (int j = 0; c + j < V; ++j) {
        float x = bfbits2f(lrow[c + j]);
        float mn = fmaxf(mx, x);
        sume = sume * __expf(mx - mn) + __expf(x - mn);
        mx = mn;
        sumx += x;
      }
    }
  }
  // merge (mx, sume) across the wave, then across the 4 waves
  {
#pragma unroll
    for (int off = 1; off < 64; off <<= 1) {
      float m2 = __shfl_xor(mx, off);
      float s2 = __shfl_xor(sume, off);
      float mn = fmaxf(mx, m2);
      sume = sume * __expf(mx - mn) + s2 * __expf(m2 - mn);
      mx = mn;
    }
    __shared__ float sm[4], ss[4];
    const int wid = threadIdx.x >> 6;
    if ((threadIdx.x & 63) == 0) {
      sm[wid] = mx;
      ss[wid] = sume;
    }
    __syncthreads();
    float M = fmaxf(fmaxf(sm[0], sm[1]), fmaxf(sm[2], sm[3]));
    sume = ss[0] * __expf(sm[0] - M) + ss[1] * __expf(sm[1] - M) +
           ss[2] * __expf(sm[2] - M) + ss[3] * __expf(sm[3] - M);
    mx = M;
    __syncthreads();
  }
  sumx = block_reduce(sumx, scratch, 1);
  const float lse = mx + __logf(sume);
  if (t == 0) {
    lse_out[row] = lse;
    const long tgt = targets[row];
    if (tgt != 0) {  // padding mask (real != 0)
      float xt = bfbits2f(lrow[tgt]);
      // (1-eps)*(lse - x_t) + eps*(lse - mean_j x_j)
      float loss = lse - (1.f - eps_ls) * xt - eps_ls * (sumx / V);
      // 256 banks: one atomic per row to a single scalar serialized all
      // 16k row-blocks (~60 us); banked contention is R/256-way.
      atomicAdd(&loss_banks[row & 255], loss * inv_batch);
    }
  }
}

// dlogits[j] = dloss/batch * mask_row * (softmax_j - (1-eps)*onehot_j - eps/V)
// dloss arrives as a device scalar so backward never synchronizes the host.
// dlogits rows carry leading dim Vp (V rounded up to 256 — the GEMM tile
// width, so TR staging of dY never reads past the allocation), with the
// pad columns written as ZERO — the downstream dX / dW GEMMs treat the
// contraction as Vp-long with no ragged-K handling (gemm_uni.hip).
__global__ __launch_bounds__(256)
void ce_bwd_kernel(const short* __restrict__ logits,
                   const long* __restrict__ targets,
                   const float* __restrict__ lse,
                   const float* __restrict__ dloss_dev,
                   short* __restrict__ dlogits,
                   long R, int V, int Vp, float inv_batch, float eps_ls) {
  const long row = blockIdx.x;
  if (row >= R) return;
  const float scale = dloss_dev[0] * inv_batch;
  const short* lrow = logits + row * V;
  short* drow = dlogits + row * Vp;
  const long tgt = targets[row];
  const int t = threadIdx.x;
  if (tgt == 0) {  // padded position contributes no gradient (Vp % 8 == 0)
    for (int c = t * 8; c < Vp; c += 256 * 8)
      *(s16x8*)(drow + c) = (s16x8){0, 0, 0, 0, 0, 0, 0, 0};
    return;
  }
  const float l = lse[row];
  const float eps_v = eps_ls / V;
  for (int c = t * 8; c < V; c += 256 * 8) {
    if (c + 8 <= V) {
      s16x8 v = *(const s16x8*)(lrow + c);
      s16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float sm = __expf(bfbits2f(v[j]) - l);
        float g = sm - eps_v - ((c + j == tgt) ? (1.f - eps_ls) : 0.f);
        o[j] = f2bfbits(g * scale);
      }
      *(s16x8*)(drow + c) = o;
    } else {
      for (int j = 0; c + j < V; ++j) {
        float sm = __expf(bfbits2f(lrow[c + j]) - l);
        float g = sm - eps_v - ((c + j == tgt) ? (1.f - eps_ls) : 0.f);
        drow[c + j] = f2bfbits(g * scale);
      }
    }
  }
  for (int c = V + t; c < Vp; c += 256) drow[c] = 0;  // zero the pad cols
}

// Fused fwd+grad: the forward pass computes (lse, loss) as ce_fwd_kernel
// and a SECOND sweep of the row (L2-hot: the row was just read by this
// block) writes the gradient scaled by 1/batch only — the autograd seed
// dloss is applied later by ce_scale_kernel, which no-ops when it is 1.0
// (every real training step).  Saves ce_bwd's separate 1.07 GB logits
// re-read per step.
__global__ __launch_bounds__(256)
void ce_fused_kernel(const short* __restrict__ logits,
                     const long* __restrict__ targets,
                     float* __restrict__ loss_banks,
                     short* __restrict__ dlogits,
                     long R, int V, int Vp, float inv_batch, float eps_ls) {
  __shared__ float scratch[4];
  const long row = blockIdx.x;
  if (row >= R) return;
  const short* lrow = logits + row * V;
  short* drow = dlogits + row * Vp;
  const int t = threadIdx.x;
  const long tgt = targets[row];

  float mx = -1e30f, sume = 0.f, sumx = 0.f;
  for (int c = t * 8; c + 8 <= V; c += 256 * 8) {
    s16x8 v = *(const s16x8*)(lrow + c);
    float x0 = bfbits2f(v[0]), x1 = bfbits2f(v[1]);
    float x2 = bfbits2f(v[2]), x3 = bfbits2f(v[3]);
    float x4 = bfbits2f(v[4]), x5 = bfbits2f(v[5]);
    float x6 = bfbits2f(v[6]), x7 = bfbits2f(v[7]);
    float cm = fmaxf(fmaxf(fmaxf(x0, x1), fmaxf(x2, x3)),
                     fmaxf(fmaxf(x4, x5), fmaxf(x6, x7)));
    float mn = fmaxf(mx, cm);
    float cs = __expf(x0 - mn) + __expf(x1 - mn) + __expf(x2 - mn) +
               __expf(x3 - mn) + __expf(x4 - mn) + __expf(x5 - mn) +
               __expf(x6 - mn) + __expf(x7 - mn);
    sume = sume * __expf(mx - mn) + cs;
    mx = mn;
    sumx += ((x0 + x1) + (x2 + x3)) + ((x4 + x5) + (x6 + x7));
  }
  {
    int c = ((V / 2048) * 2048) + t * 8;
    if (c < V && c + 8 > V) {
      for (int j = 0; c + j < V; ++j) {
        float x = bfbits2f(lrow[c + j]);
        float mn = fmaxf(mx, x);
        sume = sume * __expf(mx - mn) + __expf(x - mn);
        mx = mn;
        sumx += x;
      }
    }
  }
  {
#pragma unroll
    for (int off = 1; off < 64; off <<= 1) {
      float m2 = __shfl_xor(mx, off);
      float s2 = __shfl_xor(sume, off);
      float mn = fmaxf(mx, m2);
      sume = sume * __expf(mx - mn) + s2 * __expf(m2 - mn);
      mx = mn;
    }
    __shared__ float sm[4], ss[4];
    const int wid = threadIdx.x >> 6;
    if ((threadIdx.x & 63) == 0) {
      sm[wid] = mx;
      ss[wid] = sume;
    }
    __syncthreads();
    float M = fmaxf(fmaxf(sm[0], sm[1]), fmaxf(sm[2], sm[3]));
    sume = ss[0] * __expf(sm[0] - M) + ss[1] * __expf(sm[1] - M) +
           ss[2] * __expf(sm[2] - M) + ss[3] * __expf(sm[3] - M);
    mx = M;
    __syncthreads();
  }
  sumx = block_reduce(sumx, scratch, 1);
  const float l = mx + __logf(sume);
  if (t == 0 && tgt != 0) {
    float xt = bfbits2f(lrow[tgt]);
    float loss = l - (1.f - eps_ls) * xt - eps_ls * (sumx / V);
    atomicAdd(&loss_banks[row & 255], loss * inv_batch);
  }

  // ---- gradient sweep (rows are L2-hot from the pass above) ----------
  if (tgt == 0) {
    for (int c = t * 8; c < Vp; c += 256 * 8)
      *(s16x8*)(drow + c) = (s16x8){0, 0, 0, 0, 0, 0, 0, 0};
    return;
  }
  const float eps_v = eps_ls / V;
  for (int c = t * 8; c < V; c += 256 * 8) {
    if (c + 8 <= V) {
      s16x8 v = *(const s16x8*)(lrow + c);
      s16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float smx = __expf(bfbits2f(v[j]) - l);
        float g = smx - eps_v - ((c + j == tgt) ? (1.f - eps_ls) : 0.f);
        o[j] = f2bfbits(g * inv_batch);
      }
      *(s16x8*)(drow + c) = o;
    } else {
      for (int j = 0; c + j < V; ++j) {
        float smx = __expf(bfbits2f(lrow[c + j]) - l);
        float g = smx - eps_v - ((c + j == tgt) ? (1.f - eps_ls) : 0.f);
        drow[c + j] = f2bfbits(g * inv_batch);
      }
    }
  }
  for (int c = V + t; c < Vp; c += 256) drow[c] = 0;
}

// Apply the autograd seed AFTER the fact: no-op when *dloss == 1 (the
// training case — loss.backward() seeds 1.0), else scale in place.
__global__ void ce_scale_kernel(short* __restrict__ d, long n,
                                const float* __restrict__ dloss) {
  // grid-stride with a CAPPED grid: the no-op case (seed 1.0) must cost
  // one early-exit sweep of ~2k blocks, not the dispatch of 262k
  const float sc = dloss[0];
  if (sc == 1.0f) return;
  const long stride = (long)gridDim.x * blockDim.x * 8;
  for (long i = (long)(blockIdx.x * blockDim.x + threadIdx.x) * 8; i < n;
       i += stride) {
    if (i + 8 <= n) {
      s16x8 v = *(s16x8*)(d + i);
#pragma unroll
      for (int j = 0; j < 8; ++j) v[j] = f2bfbits(bfbits2f(v[j]) * sc);
      *(s16x8*)(d + i) = v;
    } else {
      for (long j = i; j < n; ++j) d[j] = f2bfbits(bfbits2f(d[j]) * sc);
    }
  }
}

// one wave: sum the 256 banks into the loss scalar and re-zero them for
// the next call (the bank workspace is cached per device).
__global__ void ce_loss_reduce_kernel(float* __restrict__ banks,
                                      float* __restrict__ loss_out) {
  float v = banks[threadIdx.x] + banks[threadIdx.x + 64] +
            banks[threadIdx.x + 128] + banks[threadIdx.x + 192];
  banks[threadIdx.x] = 0.f;
  banks[threadIdx.x + 64] = 0.f;
  banks[threadIdx.x + 128] = 0.f;
  banks[threadIdx.x + 192] = 0.f;
  v = wave_sum(v);
  if (threadIdx.x == 0) *loss_out = v;
}

std::vector<torch::Tensor> ce_fwd(torch::Tensor logits, torch::Tensor targets,
                                  double batch_size, double label_smoothing) {
  TORCH_CHECK(logits.is_cuda() && logits.dtype() == torch::kBFloat16 &&
              logits.dim() == 2 && logits.is_contiguous());
  TORCH_CHECK(targets.dtype() == torch::kInt64 && targets.is_contiguous());
  const long R = logits.size(0);
  const int V = logits.size(1);
  auto lse = torch::empty({R}, logits.options().dtype(torch::kFloat32));
  auto loss = torch::empty({}, logits.options().dtype(torch::kFloat32));
  static std::map<int, torch::Tensor> bank_cache;
  auto it = bank_cache.find((int)logits.get_device());
  if (it == bank_cache.end())
    it = bank_cache.emplace((int)logits.get_device(), torch::zeros(
        {256}, logits.options().dtype(torch::kFloat32))).first;
  auto banks = it->second;
  auto stream = at::hip::getCurrentHIPStream();
  ce_fwd_kernel<<<R, 256, 0, stream>>>(
      (const short*)logits.data_ptr(), targets.data_ptr<long>(),
      lse.data_ptr<float>(), banks.data_ptr<float>(), R, V,
      1.0f / (float)batch_size, (float)label_smoothing);
  ce_loss_reduce_kernel<<<1, 64, 0, stream>>>(banks.data_ptr<float>(),
                                              loss.data_ptr<float>());
  return {loss, lse};
}

// Returns (R, Vp) with Vp = V rounded up to 256 and zero pad columns;
// callers slice [:, :V] for autograd and may use the full padded tensor
// (contraction Vp) in the logits dX / dW GEMMs.
// Fused forward: returns (loss, dlogits_padded) — the gradient already
// scaled by 1/batch; ce_scale applies a non-unit autograd seed later.
std::vector<torch::Tensor> ce_fused(torch::Tensor logits,
                                    torch::Tensor targets,
                                    double batch_size,
                                    double label_smoothing) {
  TORCH_CHECK(logits.is_cuda() && logits.dtype() == torch::kBFloat16 &&
              logits.dim() == 2 && logits.is_contiguous());
  TORCH_CHECK(targets.dtype() == torch::kInt64 && targets.is_contiguous());
  const long R = logits.size(0);
  const int V = logits.size(1);
  const int Vp = (V + 255) / 256 * 256;
  auto loss = torch::empty({}, logits.options().dtype(torch::kFloat32));
  auto dlogits = torch::empty({R, Vp}, logits.options());
  static std::map<int, torch::Tensor> bank_cache;
  auto it = bank_cache.find((int)logits.get_device());
  if (it == bank_cache.end())
    it = bank_cache.emplace((int)logits.get_device(), torch::zeros(
        {256}, logits.options().dtype(torch::kFloat32))).first;
  auto banks = it->second;
  auto stream = at::hip::getCurrentHIPStream();
  ce_fused_kernel<<<R, 256, 0, stream>>>(
      (const short*)logits.data_ptr(), targets.data_ptr<long>(),
      banks.data_ptr<float>(), (short*)dlogits.data_ptr(), R, V, Vp,
      1.0f / (float)batch_size, (float)label_smoothing);
  ce_loss_reduce_kernel<<<1, 64, 0, stream>>>(banks.data_ptr<float>(),
                                              loss.data_ptr<float>());
  return {loss, dlogits};
}

void ce_scale(torch::Tensor dlogits, torch::Tensor dloss) {
  TORCH_CHECK(dlogits.is_cuda() && dlogits.dtype() == torch::kBFloat16 &&
              dlogits.is_contiguous());
  TORCH_CHECK(dloss.is_cuda() && dloss.dtype() == torch::kFloat32 &&
              dloss.numel() == 1);
  long n = dlogits.numel();
  auto stream = at::hip::getCurrentHIPStream();
  long blocks = std::min<long>(((n + 7) / 8 + 255) / 256, 2048);
  ce_scale_kernel<<<(unsigned)blocks, 256, 0, stream>>>(
      (short*)dlogits.data_ptr(), n, dloss.data_ptr<float>());
}

torch::Tensor ce_bwd(torch::Tensor logits, torch::Tensor targets,
                     torch::Tensor lse, torch::Tensor dloss,
                     double batch_size, double label_smoothing) {
  const long R = logits.size(0);
  const int V = logits.size(1);
  const int Vp = (V + 255) / 256 * 256;
  TORCH_CHECK(dloss.is_cuda() && dloss.dtype() == torch::kFloat32 &&
              dloss.numel() == 1, "dloss must be a device fp32 scalar");
  auto dlogits = torch::empty({R, Vp}, logits.options());
  auto stream = at::hip::getCurrentHIPStream();
  ce_bwd_kernel<<<R, 256, 0, stream>>>(
      (const short*)logits.data_ptr(), targets.data_ptr<long>(),
      lse.data_ptr<float>(), dloss.data_ptr<float>(),
      (short*)dlogits.data_ptr(), R, V, Vp, (float)(1.0 / batch_size),
      (float)label_smoothing);
  return dlogits;
}
