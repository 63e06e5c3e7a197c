// Decode/metric kernels: row argmax (reference train.py:111)
// and masked token accuracy (K17; SURVEY §8 Q5 — pad positions excluded).
#include "common.h"

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

// one wave per row; tracks (max, idx) pairs through the shuffle tree
__global__ __launch_bounds__(64)
void argmax_kernel(const short* __restrict__ logits, long* __restrict__ out,
                   long R, int V) {
  const long row = blockIdx.x;
  if (row >= R) return;
  const short* lrow = logits + row * V;
  const int lane = threadIdx.x;
  float best = -1e30f;
  int bidx = 0;
  for (int c = lane; c < V; c += WAVE) {
    float x = bfbits2f(lrow[c]);
    if (x > best || (x == best && c < bidx)) {
      best = x;
      bidx = c;
    }
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    float ob = __shfl_xor(best, off);
    int oi = __shfl_xor(bidx, off);
    if (ob > best || (ob == best && oi < bidx)) {
      best = ob;
      bidx = oi;
    }
  }
  if (lane == 0) out[row] = bidx;
}

__global__ __launch_bounds__(64)
void accuracy_kernel(const short* __restrict__ logits,
                     const long* __restrict__ targets,
                     unsigned long long* __restrict__ correct,
                     unsigned long long* __restrict__ total, long R, int V) {
  const long row = blockIdx.x;
  if (row >= R) return;
  const long tgt = targets[row];
  if (tgt == 0) return;  // pad position excluded
  const short* lrow = logits + row * V;
  const int lane = threadIdx.x;
  float best = -1e30f;
  int bidx = 0;
  for (int c = lane; c < V; c += WAVE) {
    float x = bfbits2f(lrow[c]);
    if (x > best || (x == best && c < bidx)) {
      best = x;
      bidx = c;
    }
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    float ob = __shfl_xor(best, off);
    int oi = __shfl_xor(bidx, off);
    if (ob > best || (ob == best && oi < bidx)) {
      best = ob;
      bidx = oi;
    }
  }
  if (lane == 0) {
    atomicAdd(total, 1ull);
    if (bidx == (int)tgt) atomicAdd(correct, 1ull);
  }
}

torch::Tensor argmax_lastdim(torch::Tensor logits) {
  TORCH_CHECK(logits.is_cuda() && logits.dim() == 2 && logits.is_contiguous());
  auto l16 = logits.dtype() == torch::kBFloat16
                 ? logits
                 : logits.to(torch::kBFloat16);
  const long R = l16.size(0);
  const int V = l16.size(1);
  auto out = torch::empty({R}, l16.options().dtype(torch::kInt64));
  auto stream = at::hip::getCurrentHIPStream();
  argmax_kernel<<<R, 64, 0, stream>>>((const short*)l16.data_ptr(),
                                      out.data_ptr<long>(), R, V);
  return out;
}

std::vector<int64_t> accuracy(torch::Tensor logits, torch::Tensor targets) {
  TORCH_CHECK(logits.is_cuda() && logits.dim() == 2 && logits.is_contiguous());
  auto l16 = logits.dtype() == torch::kBFloat16
                 ? logits
                 : logits.to(torch::kBFloat16);
  const long R = l16.size(0);
  const int V = l16.size(1);
  auto counters =
      torch::zeros({2}, l16.options().dtype(torch::kInt64));
  auto stream = at::hip::getCurrentHIPStream();
  accuracy_kernel<<<R, 64, 0, stream>>>(
      (const short*)l16.data_ptr(), targets.data_ptr<long>(),
      (unsigned long long*)counters.data_ptr<int64_t>(),
      (unsigned long long*)(counters.data_ptr<int64_t>() + 1), R, V);
  auto cpu = counters.cpu();
  return {cpu[0].item<int64_t>(), cpu[1].item<int64_t>()};
}
