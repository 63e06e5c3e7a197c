// Fused embedding-gather * sqrt(d) + positional-encoding add (SURVEY.md
// K10; semantics = reference Encoder.py:51-53).  Backward scatter-adds
// dW[tok] += dy * sqrt(d) into an fp32 workspace (atomics), then casts.
#include "common.h"

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

__global__ __launch_bounds__(256)
void embed_pe_fwd_kernel(const long* __restrict__ tokens,
                         const short* __restrict__ weight,
                         const short* __restrict__ pe,
                         short* __restrict__ y, int S, int D, long n_rows,
                         float scale) {
  const long row = blockIdx.x * 4 + (threadIdx.x >> 6);
  if (row >= n_rows) return;
  const int lane = threadIdx.x & 63;
  const int s = row % S;  // position within the sequence
  const long tok = tokens[row];
  const short* wrow = weight + tok * D;
  const short* perow = pe + (long)s * D;
  short* yrow = y + row * D;
  for (int c = lane * 8; c < D; c += WAVE * 8) {
    if (c + 8 <= D) {
      s16x8 wv = *(const s16x8*)(wrow + c);
      s16x8 pv = *(const s16x8*)(perow + c);
      s16x8 yv;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        yv[j] = f2bfbits(bfbits2f(wv[j]) * scale + bfbits2f(pv[j]));
      *(s16x8*)(yrow + c) = yv;
    } else {
      for (int j = 0; c + j < D; ++j)
        yrow[c + j] =
            f2bfbits(bfbits2f(wrow[c + j]) * scale + bfbits2f(perow[c + j]));
    }
  }
}

__global__ __launch_bounds__(256)
void embed_pe_bwd_kernel(const short* __restrict__ dy,
                         const long* __restrict__ tokens,
                         float* __restrict__ dw_f32, int D, long n_rows,
                         float scale) {
  const long row = blockIdx.x * 4 + (threadIdx.x >> 6);
  if (row >= n_rows) return;
  const int lane = threadIdx.x & 63;
  const long tok = tokens[row];
  float* drow = dw_f32 + tok * D;
  const short* dyrow = dy + row * D;
  for (int c = lane; c < D; c += WAVE)
    atomicAdd(&drow[c], bfbits2f(dyrow[c]) * scale);
}

__global__ void cast_f32_bf16_kernel(const float* __restrict__ in,
                                     short* __restrict__ out, long n) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) out[i] = f2bfbits(in[i]);
}

torch::Tensor embed_pe_fwd(torch::Tensor tokens, torch::Tensor weight,
                           torch::Tensor pe) {
  TORCH_CHECK(tokens.is_cuda() && tokens.dtype() == torch::kInt64 &&
              tokens.dim() == 2 && tokens.is_contiguous());
  TORCH_CHECK(weight.dtype() == torch::kBFloat16 && weight.is_contiguous());
  TORCH_CHECK(pe.dtype() == torch::kBFloat16 && pe.is_contiguous());
  const int B = tokens.size(0), S = tokens.size(1), D = weight.size(1);
  TORCH_CHECK(pe.size(0) >= S, "PE table shorter than sequence");
  auto y = torch::empty({B, S, D}, weight.options());
  long n_rows = (long)B * S;
  auto stream = at::hip::getCurrentHIPStream();
  embed_pe_fwd_kernel<<<cdiv(n_rows, 4), 256, 0, stream>>>(
      tokens.data_ptr<long>(), (const short*)weight.data_ptr(),
      (const short*)pe.data_ptr(), (short*)y.data_ptr(), S, D, n_rows,
      sqrtf((float)D));
  return y;
}

torch::Tensor embed_pe_bwd(torch::Tensor dy, torch::Tensor tokens,
                           int64_t vocab,
                           c10::optional<torch::Tensor> out_opt) {
  const int D = dy.size(-1);
  long n_rows = dy.numel() / D;
  auto dw_f = torch::zeros({vocab, (long)D},
                           dy.options().dtype(torch::kFloat32));
  auto stream = at::hip::getCurrentHIPStream();
  embed_pe_bwd_kernel<<<cdiv(n_rows, 4), 256, 0, stream>>>(
      (const short*)dy.data_ptr(), tokens.data_ptr<long>(),
      dw_f.data_ptr<float>(), D, n_rows, sqrtf((float)D));
  torch::Tensor dw;
  if (out_opt.has_value()) {
    dw = *out_opt;
    TORCH_CHECK(dw.is_cuda() && dw.dtype() == torch::kBFloat16 &&
                dw.is_contiguous() && dw.numel() == vocab * (long)D);
  } else {
    dw = torch::empty({vocab, (long)D}, dy.options());
  }
  long n = vocab * (long)D;
  cast_f32_bf16_kernel<<<(n + 1023) / 1024, 1024, 0, stream>>>(
      dw_f.data_ptr<float>(), (short*)dw.data_ptr(), n);
  return dw;
}
