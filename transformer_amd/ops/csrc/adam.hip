// Fused flat-buffer Adam (reference train.py:65-66:
// β1=0.9, β2=0.98, ε=1e-9, Noam LR computed host-side per step).
// One kernel over the whole flat parameter set: fp32 master weights + m/v,
// bf16 gradient in, bf16 parameter copy out (the model's live weights).
#include "common.h"

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

// Vectorized: 28 B/param of HBM traffic (fp32 master/m/v RW + bf16 g/p),
// f32x4 + s16x4 transactions so the kernel runs at memory speed-of-light.
__global__ void adam_kernel(float* __restrict__ master, float* __restrict__ m,
                            float* __restrict__ v,
                            const short* __restrict__ grad,
                            short* __restrict__ param, long n, float lr,
                            float b1, float b2, float eps, float bc1,
                            float bc2) {
  long i = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 4;
  if (i + 4 <= n) {
    f32x4 mm = *(const f32x4*)(m + i);
    f32x4 vv = *(const f32x4*)(v + i);
    f32x4 ww = *(const f32x4*)(master + i);
    s16x4 g4 = *(const s16x4*)(grad + i);
    s16x4 p4;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float g = bfbits2f(g4[j]);
      float mj = b1 * mm[j] + (1.f - b1) * g;
      float vj = b2 * vv[j] + (1.f - b2) * g * g;
      mm[j] = mj;
      vv[j] = vj;
      float w = ww[j] - lr * (mj * bc1) / (sqrtf(vj * bc2) + eps);
      ww[j] = w;
      p4[j] = f2bfbits(w);
    }
    *(f32x4*)(m + i) = mm;
    *(f32x4*)(v + i) = vv;
    *(f32x4*)(master + i) = ww;
    *(s16x4*)(param + i) = p4;
  } else {
    for (long j = i; j < n; ++j) {
      float g = bfbits2f(grad[j]);
      float mj = b1 * m[j] + (1.f - b1) * g;
      float vj = b2 * v[j] + (1.f - b2) * g * g;
      m[j] = mj;
      v[j] = vj;
      float w = master[j] - lr * (mj * bc1) / (sqrtf(vj * bc2) + eps);
      master[j] = w;
      param[j] = f2bfbits(w);
    }
  }
}

// Device-side schedule prelude for HIP-graph capture (Q12): ONE thread
// increments the step tensor and derives {lr, bc1, bc2} so a captured
// graph replays with a advancing Noam schedule instead of baked host
// constants.  lr = d_model^-0.5 * min(step^-0.5, step * warmup^-1.5).
__global__ void adam_coefs_kernel(long long* __restrict__ step,
                                  float* __restrict__ coefs, float dmr,
                                  float warmup_pow, float b1, float b2) {
  long long s = *step + 1;
  *step = s;
  float fs = (float)s;
  float lr = dmr * fminf(rsqrtf(fs), fs * warmup_pow);
  coefs[0] = lr;
  coefs[1] = 1.0f / (1.0f - powf(b1, fs));
  coefs[2] = 1.0f / (1.0f - powf(b2, fs));
}

// Variant of adam_kernel whose lr/bias-corrections come from the coefs
// buffer written by adam_coefs_kernel (graph-capturable).
__global__ void adam_dev_kernel(float* __restrict__ master,
                                float* __restrict__ m, float* __restrict__ v,
                                const short* __restrict__ grad,
                                short* __restrict__ param, long n,
                                const float* __restrict__ coefs, float b1,
                                float b2, float eps) {
  const float lr = coefs[0], bc1 = coefs[1], bc2 = coefs[2];
  long i = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 4;
  if (i + 4 <= n) {
    f32x4 mm = *(const f32x4*)(m + i);
    f32x4 vv = *(const f32x4*)(v + i);
    f32x4 ww = *(const f32x4*)(master + i);
    s16x4 g4 = *(const s16x4*)(grad + i);
    s16x4 p4;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float g = bfbits2f(g4[j]);
      float mj = b1 * mm[j] + (1.f - b1) * g;
      float vj = b2 * vv[j] + (1.f - b2) * g * g;
      mm[j] = mj;
      vv[j] = vj;
      float w = ww[j] - lr * (mj * bc1) / (sqrtf(vj * bc2) + eps);
      ww[j] = w;
      p4[j] = f2bfbits(w);
    }
    *(f32x4*)(m + i) = mm;
    *(f32x4*)(v + i) = vv;
    *(f32x4*)(master + i) = ww;
    *(s16x4*)(param + i) = p4;
  } else {
    for (long j = i; j < n; ++j) {
      float g = bfbits2f(grad[j]);
      float mj = b1 * m[j] + (1.f - b1) * g;
      float vj = b2 * v[j] + (1.f - b2) * g * g;
      m[j] = mj;
      v[j] = vj;
      float w = master[j] - lr * (mj * bc1) / (sqrtf(vj * bc2) + eps);
      master[j] = w;
      param[j] = f2bfbits(w);
    }
  }
}

// Graph-capturable fused Adam: advances `step` (int64[1], device) and runs
// the update with the device-derived Noam lr.  d_model/warmup fixed.
void adam_fused_dev(torch::Tensor master, torch::Tensor m, torch::Tensor v,
                    torch::Tensor grad, torch::Tensor param,
                    torch::Tensor step, torch::Tensor coefs, double d_model,
                    double warmup, double beta1, double beta2, double eps) {
  TORCH_CHECK(master.dtype() == torch::kFloat32 && master.is_contiguous());
  TORCH_CHECK(step.dtype() == torch::kInt64 && step.numel() == 1 &&
              step.is_cuda());
  TORCH_CHECK(coefs.dtype() == torch::kFloat32 && coefs.numel() >= 3 &&
              coefs.is_cuda());
  long n = master.numel();
  auto stream = at::hip::getCurrentHIPStream();
  adam_coefs_kernel<<<1, 1, 0, stream>>>(
      (long long*)step.data_ptr(), coefs.data_ptr<float>(),
      1.0f / sqrtf((float)d_model),
      powf((float)warmup, -1.5f), (float)beta1, (float)beta2);
  adam_dev_kernel<<<((n + 3) / 4 + 255) / 256, 256, 0, stream>>>(
      master.data_ptr<float>(), m.data_ptr<float>(), v.data_ptr<float>(),
      (const short*)grad.data_ptr(), (short*)param.data_ptr(), n,
      coefs.data_ptr<float>(), (float)beta1, (float)beta2, (float)eps);
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
