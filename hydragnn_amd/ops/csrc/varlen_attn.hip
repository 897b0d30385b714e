// Segment-varlen attention for GPS global attention (gfx950).
//
// Graph batches are many variable-size segments (molecules 10-128
// nodes, OC20 slabs ~80, supercells 1k+); the dense-batch SDPA path
// pads every graph to max_N and runs masked attention over padding.
// This kernel processes one (graph, head) per workgroup with K and V
// staged through LDS in tiles, one thread per query row, and an
// online softmax — no padding, no masks, no [B, maxN, ...]
// materialization, and no segment-length limit (r2: the r1 kernel
// required seg <= 256 and fp32 only).
//
// dtype: fp32 or bf16 inputs; LDS staging and accumulation in fp32.
// head_dim <= 64.  Backward is recompute-based on the Python side
// (torch ops), keeping double-backward support without a
// hand-written second-order kernel.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

namespace {

constexpr int kMaxDh = 64;
constexpr int kTile = 128;      // K/V rows staged per LDS pass
constexpr int kThreads = 128;

template <typename T>
__device__ inline float to_f32(T v);
template <>
__device__ inline float to_f32<float>(float v) { return v; }
template <>
__device__ inline float to_f32<__hip_bfloat16>(__hip_bfloat16 v) {
  return __bfloat162float(v);
}
template <typename T>
__device__ inline T from_f32(float v);
template <>
__device__ inline float from_f32<float>(float v) { return v; }
template <>
__device__ inline __hip_bfloat16 from_f32<__hip_bfloat16>(float v) {
  return __float2bfloat16(v);
}

template <typename T>
__global__ __launch_bounds__(kThreads) void varlen_attn_kernel(
    const T* __restrict__ Q,   // [N, H, dh]
    const T* __restrict__ K,
    const T* __restrict__ V,
    T* __restrict__ O,         // [N, H, dh]
    const long* __restrict__ ptr,  // [G+1] node offsets per graph
    int H, int dh, float scale) {
  __shared__ float lK[kTile * kMaxDh];
  __shared__ float lV[kTile * kMaxDh];
  // per-thread state slices (runtime-indexed; LDS not registers)
  __shared__ float lAcc[kThreads * (kMaxDh + 1)];
  __shared__ float lQ[kThreads * (kMaxDh + 1)];

  int g = blockIdx.x;
  int h = blockIdx.y;
  long lo = ptr[g], hi = ptr[g + 1];
  int n = (int)(hi - lo);
  if (n <= 0) return;

  float* acc = &lAcc[threadIdx.x * (kMaxDh + 1)];
  float* qv = &lQ[threadIdx.x * (kMaxDh + 1)];

  // queries handled by this thread: q0, q0+kThreads, ...  To keep the
  // tile loop uniform across the block (needed for __syncthreads),
  // every thread walks all tiles even when it has no query row.
  for (int q0 = 0; q0 < n; q0 += kThreads) {
    int q = q0 + threadIdx.x;
    bool hasq = q < n;
    float m = -1e30f, l = 0.f;
    if (hasq) {
      const T* qp = &Q[((lo + q) * (long)H + h) * dh];
      for (int d = 0; d < dh; ++d) qv[d] = to_f32<T>(qp[d]);
      for (int d = 0; d < dh; ++d) acc[d] = 0.f;
    }
    for (int t0 = 0; t0 < n; t0 += kTile) {
      int tn = n - t0 < kTile ? n - t0 : kTile;
      __syncthreads();
      for (int i = threadIdx.x; i < tn * dh; i += kThreads) {
        int row = i / dh, d = i - (i / dh) * dh;
        lK[row * kMaxDh + d] =
            to_f32<T>(K[((lo + t0 + row) * (long)H + h) * dh + d]);
        lV[row * kMaxDh + d] =
            to_f32<T>(V[((lo + t0 + row) * (long)H + h) * dh + d]);
      }
      __syncthreads();
      if (hasq) {
        for (int k = 0; k < tn; ++k) {
          float s = 0.f;
          const float* kp = &lK[k * kMaxDh];
          for (int d = 0; d < dh; ++d) s += qv[d] * kp[d];
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
      }
    }
    if (hasq) {
      float inv = 1.f / l;
      T* op = &O[((lo + q) * (long)H + h) * dh];
      for (int d = 0; d < dh; ++d) op[d] = from_f32<T>(acc[d] * inv);
    }
    __syncthreads();
  }
}

}  // namespace

torch::Tensor varlen_attention(torch::Tensor Q, torch::Tensor K,
                               torch::Tensor V, torch::Tensor ptr) {
  TORCH_CHECK(Q.is_cuda() && Q.is_contiguous());
  TORCH_CHECK(Q.scalar_type() == at::ScalarType::Float ||
              Q.scalar_type() == at::ScalarType::BFloat16,
              "varlen_attention expects fp32 or bf16 q/k/v");
  TORCH_CHECK(K.scalar_type() == Q.scalar_type() &&
              V.scalar_type() == Q.scalar_type());
  long N = Q.size(0);
  int H = Q.size(1), dh = Q.size(2);
  TORCH_CHECK(dh <= kMaxDh, "head_dim must be <= 64");
  long G = ptr.numel() - 1;
  auto O = torch::empty_like(Q);
  if (N == 0 || G == 0) return O;
  float scale = 1.0f / std::sqrt((float)dh);
  auto stream = at::hip::getCurrentHIPStream().stream();
  auto p = ptr.contiguous();
  if (Q.scalar_type() == at::ScalarType::Float) {
    hipLaunchKernelGGL((varlen_attn_kernel<float>), dim3(G, H),
                       dim3(kThreads), 0, stream, Q.data_ptr<float>(),
                       K.data_ptr<float>(), V.data_ptr<float>(),
                       O.data_ptr<float>(), p.data_ptr<long>(), H, dh,
                       scale);
  } else {
    hipLaunchKernelGGL((varlen_attn_kernel<__hip_bfloat16>), dim3(G, H),
                       dim3(kThreads), 0, stream,
                       reinterpret_cast<const __hip_bfloat16*>(Q.data_ptr()),
                       reinterpret_cast<const __hip_bfloat16*>(K.data_ptr()),
                       reinterpret_cast<const __hip_bfloat16*>(V.data_ptr()),
                       reinterpret_cast<__hip_bfloat16*>(O.data_ptr()),
                       p.data_ptr<long>(), H, dh, scale);
  }
  return O;
}
