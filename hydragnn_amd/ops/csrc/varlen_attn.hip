// Segment-varlen attention for GPS global attention (gfx950).
//
// Graph batches are many SMALL segments (molecules: 10-128 nodes); the
// dense-batch SDPA path pads every graph to max_N and runs masked
// attention over the padding.  This kernel processes one (graph, head)
// per block with K and V staged in LDS, one thread per query row, and
// an online softmax — no padding, no masks, no [B, maxN, ...]
// materialization.
//
// Scope: head_dim <= 32, segment length <= kMaxSeg (LDS-bound).
// Backward is recompute-based on the Python side (torch ops), keeping
// double-backward support without a hand-written second-order kernel.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>

namespace {

constexpr int kMaxDh = 32;
constexpr int kMaxSeg = 256;
constexpr int kThreads = 128;

__global__ void varlen_attn_kernel(
    const float* __restrict__ Q,   // [N, H, dh]
    const float* __restrict__ K,
    const float* __restrict__ V,
    float* __restrict__ O,         // [N, H, dh]
    const long* __restrict__ ptr,  // [G+1] node offsets per graph
    int H, int dh, float scale) {
  __shared__ float lK[kMaxSeg * kMaxDh];
  __shared__ float lV[kMaxSeg * kMaxDh];
  // per-thread accumulator slice (avoids runtime-indexed registers)
  __shared__ float lAcc[kThreads * (kMaxDh + 1)];

  int g = blockIdx.x;
  int h = blockIdx.y;
  long lo = ptr[g], hi = ptr[g + 1];
  int n = (int)(hi - lo);
  if (n <= 0) return;

  // stage K and V for this (graph, head)
  for (int i = threadIdx.x; i < n * dh; i += blockDim.x) {
    int row = i / dh, d = i - (i / dh) * dh;
    lK[row * kMaxDh + d] = K[((lo + row) * H + h) * dh + d];
    lV[row * kMaxDh + d] = V[((lo + row) * H + h) * dh + d];
  }
  __syncthreads();

  float* acc = &lAcc[threadIdx.x * (kMaxDh + 1)];
  for (int q = threadIdx.x; q < n; q += blockDim.x) {
    const float* qp = &Q[((lo + q) * H + h) * dh];
    float m = -1e30f, l = 0.f;
    for (int d = 0; d < dh; ++d) acc[d] = 0.f;
    for (int k = 0; k < n; ++k) {
      float s = 0.f;
      const float* kp = &lK[k * kMaxDh];
      for (int d = 0; d < dh; ++d) s += qp[d] * kp[d];
      s *= scale;
      float m_new = s > m ? s : m;
      float alpha = __expf(m - m_new);
      float p = __expf(s - m_new);
      l = l * alpha + p;
      const float* vp = &lV[k * kMaxDh];
      for (int d = 0; d < dh; ++d)
        acc[d] = acc[d] * alpha + p * vp[d];
      m = m_new;
    }
    float inv = 1.f / l;
    float* op = &O[((lo + q) * H + h) * dh];
    for (int d = 0; d < dh; ++d) op[d] = acc[d] * inv;
  }
}

}  // namespace

torch::Tensor varlen_attention(torch::Tensor Q, torch::Tensor K,
                               torch::Tensor V, torch::Tensor ptr) {
  TORCH_CHECK(Q.is_cuda() && Q.is_contiguous());
  TORCH_CHECK(Q.scalar_type() == at::ScalarType::Float,
              "varlen_attention expects fp32 q/k/v");
  long N = Q.size(0);
  int H = Q.size(1), dh = Q.size(2);
  TORCH_CHECK(dh <= kMaxDh, "head_dim must be <= 32");
  long G = ptr.numel() - 1;
  auto O = torch::empty_like(Q);
  if (N == 0 || G == 0) return O;
  float scale = 1.0f / std::sqrt((float)dh);
  auto stream = at::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(varlen_attn_kernel, dim3(G, H), dim3(kThreads), 0,
                     stream, Q.data_ptr<float>(), K.data_ptr<float>(),
                     V.data_ptr<float>(), O.data_ptr<float>(),
                     ptr.contiguous().data_ptr<long>(), H, dh, scale);
  return O;
}
