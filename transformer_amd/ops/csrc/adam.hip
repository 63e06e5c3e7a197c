// Fused flat-buffer Adam (reference train.py:65-66:
// β1=0.9, β2=0.98, ε=1e-9, Noam LR computed host-side per step).
// One kernel over the whole flat parameter set: fp32 master weights + m/v,
// bf16 gradient in, bf16 parameter copy out (the model's live weights).
#include "common.h"

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

__global__ void adam_kernel(float* __restrict__ master, float* __restrict__ m,
                            float* __restrict__ v,
                            const short* __restrict__ grad,
                            short* __restrict__ param, long n, float lr,
                            float b1, float b2, float eps, float bc1,
                            float bc2) {
  long i = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 4;
  if (i >= n) return;
  int cnt = min(4l, n - i);
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    if (j >= cnt) break;
    float g = bfbits2f(grad[i + j]);
    float mj = b1 * m[i + j] + (1.f - b1) * g;
    float vj = b2 * v[i + j] + (1.f - b2) * g * g;
    m[i + j] = mj;
    v[i + j] = vj;
    float upd = lr * (mj * bc1) / (sqrtf(vj * bc2) + eps);
    float w = master[i + j] - upd;
    master[i + j] = w;
    param[i + j] = f2bfbits(w);
  }
}

void adam_fused(torch::Tensor master, torch::Tensor m, torch::Tensor v,
                torch::Tensor grad, torch::Tensor param, double lr,
                double beta1, double beta2, double eps, int64_t step) {
  TORCH_CHECK(master.dtype() == torch::kFloat32 && master.is_contiguous());
  TORCH_CHECK(grad.dtype() == torch::kBFloat16 && param.dtype() == torch::kBFloat16);
  long n = master.numel();
  TORCH_CHECK(m.numel() == n && v.numel() == n && grad.numel() == n &&
              param.numel() == n);
  float bc1 = 1.0f / (1.0f - powf((float)beta1, (float)step));
  float bc2 = 1.0f / (1.0f - powf((float)beta2, (float)step));
  auto stream = at::hip::getCurrentHIPStream();
  adam_kernel<<<((n + 3) / 4 + 255) / 256, 256, 0, stream>>>(
      master.data_ptr<float>(), m.data_ptr<float>(), v.data_ptr<float>(),
      (const short*)grad.data_ptr(), (short*)param.data_ptr(), n, (float)lr,
      (float)beta1, (float)beta2, (float)eps, bc1, bc2);
}
