// Narrow-output linear (GEMV family) for head readout layers (gfx950).
//
// The decoder heads end in Linear(hidden, out) with out in [1, 8)
// over M ~ 1e4-1e6 rows (reference heads:
// /root/reference/hydragnn/models/Base.py readout stacks).  hipBLASLt
// schedules these N=1 shapes on MT1x4x256 tiles at ~170 us; the op is
// memory-bound (read A once) and belongs at the HBM roofline (~5 us).
//
// Mapping: 8 lanes per row (each lane reads 8 contiguous bf16 = 16 B,
// so a wave covers 8 rows at full coalescing); W[N, K] staged in LDS;
// per-lane fp32 dot over its K/8 slice, 3-step __shfl_xor reduction
// across the 8 lanes of the row, lane 0 writes all N outputs.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

namespace {

using bf16x8g = __attribute__((ext_vector_type(8))) __bf16;

constexpr int MAX_N = 8;
constexpr int THREADS = 256;
constexpr int LANES_PER_ROW = 8;
constexpr int ROWS_PER_BLOCK = THREADS / LANES_PER_ROW;  // 32

__global__ __launch_bounds__(THREADS) void gemv_small_n_kernel(
    const __hip_bfloat16* __restrict__ A,   // [M, K]
    const __hip_bfloat16* __restrict__ W,   // [N, K]
    const float* __restrict__ bias,         // [N] or nullptr
    __hip_bfloat16* __restrict__ C,         // [M, N]
    long M, int N, int K) {
  extern __shared__ __hip_bfloat16 lW[];    // [N, K]
  for (int i = threadIdx.x; i < N * K; i += THREADS)
    lW[i] = W[i];
  __syncthreads();

  const int tid = threadIdx.x;
  const int sub = tid & (LANES_PER_ROW - 1);   // lane within row
  const long row = (long)blockIdx.x * ROWS_PER_BLOCK
                   + (tid / LANES_PER_ROW);
  if (row >= M) return;

  float acc[MAX_N];
  for (int n = 0; n < MAX_N; ++n) acc[n] = 0.f;

  // K-slices of 8 bf16 per lane; stride LANES_PER_ROW*8 = 64
  for (int k0 = sub * 8; k0 < K; k0 += LANES_PER_ROW * 8) {
    bf16x8g a = *reinterpret_cast<const bf16x8g*>(A + row * K + k0);
    for (int n = 0; n < N; ++n) {
      bf16x8g w = *reinterpret_cast<const bf16x8g*>(&lW[n * K + k0]);
      float s = 0.f;
      for (int q = 0; q < 8; ++q) s += (float)a[q] * (float)w[q];
      acc[n] += s;
    }
  }
  // reduce across the 8 lanes of this row
  for (int off = 1; off < LANES_PER_ROW; off <<= 1)
    for (int n = 0; n < N; ++n)
      acc[n] += __shfl_xor(acc[n], off, 64);
  if (sub == 0) {
    for (int n = 0; n < N; ++n) {
      float b = bias != nullptr ? bias[n] : 0.f;
      C[row * N + n] = __float2bfloat16(acc[n] + b);
    }
  }
}

}  // namespace

torch::Tensor gemv_small_n(torch::Tensor A, torch::Tensor W,
                           c10::optional<torch::Tensor> bias) {
  TORCH_CHECK(A.is_cuda() && A.is_contiguous());
  TORCH_CHECK(W.is_cuda() && W.is_contiguous());
  TORCH_CHECK(A.scalar_type() == at::ScalarType::BFloat16);
  TORCH_CHECK(W.scalar_type() == at::ScalarType::BFloat16);
  long M = A.size(0);
  int K = A.size(1);
  int N = W.size(0);
  TORCH_CHECK(W.size(1) == K, "inner dims mismatch");
  TORCH_CHECK(N >= 1 && N <= MAX_N, "gemv_small_n needs 1 <= N <= 8");
  TORCH_CHECK(K % 8 == 0, "gemv_small_n needs K % 8 == 0");
  auto C = torch::empty({M, (long)N}, A.options());
  if (M == 0) return C;
  const float* bias_ptr = nullptr;
  torch::Tensor bias_f;
  if (bias.has_value()) {
    bias_f = bias->to(torch::kFloat).contiguous();
    bias_ptr = bias_f.data_ptr<float>();
  }
  long blocks = (M + ROWS_PER_BLOCK - 1) / ROWS_PER_BLOCK;
  size_t lds = (size_t)N * K * sizeof(__hip_bfloat16);
  auto stream = at::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(gemv_small_n_kernel, dim3(blocks), dim3(THREADS),
                     lds, stream,
                     reinterpret_cast<const __hip_bfloat16*>(A.data_ptr()),
                     reinterpret_cast<const __hip_bfloat16*>(W.data_ptr()),
                     bias_ptr,
                     reinterpret_cast<__hip_bfloat16*>(C.data_ptr()),
                     M, N, K);
  return C;
}
