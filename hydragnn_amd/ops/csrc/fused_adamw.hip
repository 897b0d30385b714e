// Fused AdamW over a flat fp32 parameter/gradient buffer (gfx950).
//
// The captured train step (train/captured.py) keeps every gradient as
// a view into ONE flat buffer; flattening the parameters the same way
// turns the optimizer into a single elementwise kernel instead of
// ~10 foreach launches — and because `step` is carried in a device
// tensor incremented by a stream-ordered prelude kernel, the whole
// update is hipGraph-capturable (the graph replays the optimizer too).
//
// Math matches torch.optim.AdamW (decoupled weight decay, bias
// correction):
//   p *= 1 - lr*wd
//   m = b1*m + (1-b1)*g ; v = b2*v + (1-b2)*g^2
//   p -= lr * (m/(1-b1^t)) / (sqrt(v/(1-b2^t)) + eps)

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>

namespace {

__global__ void adamw_step_inc_kernel(float* __restrict__ step) {
  if (threadIdx.x == 0 && blockIdx.x == 0) step[0] += 1.0f;
}

__global__ void fused_adamw_kernel(
    float* __restrict__ p, const float* __restrict__ g,
    float* __restrict__ m, float* __restrict__ v,
    const float* __restrict__ step,  // [1], already incremented
    long n, float lr, float beta1, float beta2, float eps, float wd) {
  const float t = step[0];
  const float bc1 = 1.0f - __powf(beta1, t);
  const float bc2 = 1.0f - __powf(beta2, t);
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    float gi = g[i];
    float pi = p[i] * (1.0f - lr * wd);
    float mi = beta1 * m[i] + (1.0f - beta1) * gi;
    float vi = beta2 * v[i] + (1.0f - beta2) * gi * gi;
    m[i] = mi;
    v[i] = vi;
    float denom = __fsqrt_rn(vi / bc2) + eps;
    p[i] = pi - lr * (mi / bc1) / denom;
  }
}

}  // namespace

void fused_adamw(torch::Tensor p, torch::Tensor g, torch::Tensor m,
                 torch::Tensor v, torch::Tensor step, double lr,
                 double beta1, double beta2, double eps, double wd) {
  TORCH_CHECK(p.is_cuda() && p.is_contiguous()
              && p.scalar_type() == at::ScalarType::Float);
  TORCH_CHECK(g.sizes() == p.sizes() && m.sizes() == p.sizes()
              && v.sizes() == p.sizes());
  TORCH_CHECK(step.numel() == 1
              && step.scalar_type() == at::ScalarType::Float);
  long n = p.numel();
  int block = 256;
  long blocks = std::min((n + block - 1) / block, (long)8192);
  auto stream = at::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(adamw_step_inc_kernel, dim3(1), dim3(64), 0,
                     stream, step.data_ptr<float>());
  hipLaunchKernelGGL(fused_adamw_kernel, dim3(blocks), dim3(block), 0,
                     stream, p.data_ptr<float>(), g.data_ptr<float>(),
                     m.data_ptr<float>(), v.data_ptr<float>(),
                     step.data_ptr<float>(), n, (float)lr,
                     (float)beta1, (float)beta2, (float)eps,
                     (float)wd);
}


namespace {

// Mixed-precision variant: bf16 working params/grads, fp32 master +
// moments (pure-bf16 compute with fp32 optimizer state).
__global__ void fused_adamw_bf16_kernel(
    __hip_bfloat16* __restrict__ p, const __hip_bfloat16* __restrict__ g,
    float* __restrict__ master, float* __restrict__ m,
    float* __restrict__ v, const float* __restrict__ step,
    long n, float lr, float beta1, float beta2, float eps, float wd) {
  const float t = step[0];
  const float bc1 = 1.0f - __powf(beta1, t);
  const float bc2 = 1.0f - __powf(beta2, t);
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    float gi = __bfloat162float(g[i]);
    float pi = master[i] * (1.0f - lr * wd);
    float mi = beta1 * m[i] + (1.0f - beta1) * gi;
    float vi = beta2 * v[i] + (1.0f - beta2) * gi * gi;
    m[i] = mi;
    v[i] = vi;
    float denom = __fsqrt_rn(vi / bc2) + eps;
    pi -= lr * (mi / bc1) / denom;
    master[i] = pi;
    p[i] = __float2bfloat16(pi);
  }
}

}  // namespace

void fused_adamw_bf16(torch::Tensor p, torch::Tensor g,
                      torch::Tensor master, torch::Tensor m,
                      torch::Tensor v, torch::Tensor step, double lr,
                      double beta1, double beta2, double eps,
                      double wd) {
  TORCH_CHECK(p.is_cuda() && p.is_contiguous()
              && p.scalar_type() == at::ScalarType::BFloat16);
  TORCH_CHECK(g.scalar_type() == at::ScalarType::BFloat16);
  TORCH_CHECK(master.scalar_type() == at::ScalarType::Float);
  long n = p.numel();
  int block = 256;
  long blocks = std::min((n + block - 1) / block, (long)8192);
  auto stream = at::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(adamw_step_inc_kernel, dim3(1), dim3(64), 0,
                     stream, step.data_ptr<float>());
  hipLaunchKernelGGL(
      fused_adamw_bf16_kernel, dim3(blocks), dim3(block), 0, stream,
      reinterpret_cast<__hip_bfloat16*>(p.data_ptr()),
      reinterpret_cast<const __hip_bfloat16*>(g.data_ptr()),
      master.data_ptr<float>(), m.data_ptr<float>(),
      v.data_ptr<float>(), step.data_ptr<float>(), n, (float)lr,
      (float)beta1, (float)beta2, (float)eps, (float)wd);
}
