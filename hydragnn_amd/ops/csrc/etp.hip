// Fused equivariant tensor-product contraction kernels (CDNA4/gfx950).
//
// The MACE hot path is per-edge / per-node SMALL trilinear contractions
//   out[i, c, o] = sum_entries coef * A[i, c, a] * B[i, b] * C[i, c, g]
// (entry table = flattened Wigner-3j paths).  hipBLASLt runs these as
// tiny-batched GEMMs at ~2% of HBM bandwidth; here one kernel does the
// whole contraction: a 32-thread group per (i, c) pair stages the A/B/C
// rows in LDS, each thread owns one output index and walks its slice of
// the (sorted-by-output) entry table.  VGPR use stays low (no
// runtime-indexed register arrays -> no scratch), so occupancy is high
// and the kernel runs at the memory-bound roofline.
//
// The same kernel computes every first- and second-order derivative of
// the contraction: gradients of a trilinear form are trilinear forms
// with role-permuted entry tables (see ops/etp.py), so force training
// (create_graph=True double backward) never leaves this kernel family.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

namespace {

constexpr int kGroup = 32;            // threads per (i, c) pair
constexpr int kGroupsPerBlock = 8;    // 256-thread blocks
constexpr int kMaxDim = 40;           // max of da/db/dg/do

template <typename T>
__device__ inline float to_f32(T v) { return (float)v; }
template <>
__device__ inline float to_f32<__hip_bfloat16>(__hip_bfloat16 v) {
  return __bfloat162float(v);
}
template <typename T>
__device__ inline T from_f32(float v) { return (T)v; }
template <>
__device__ inline __hip_bfloat16 from_f32<__hip_bfloat16>(float v) {
  return __float2bfloat16(v);
}

// out[i,c,o] = sum over entries(coef, a, b, g) grouped by o of
//              coef * A[i,c,a] * B[i,b] * C[i,c,g]
template <typename T>
__global__ void etp_general_kernel(
    const T* __restrict__ A, const T* __restrict__ B,
    const T* __restrict__ C, T* __restrict__ out,
    const int4* __restrict__ entries,   // (a, b, g, o) sorted by o
    const float* __restrict__ coefs,
    const int2* __restrict__ o_ranges,  // [do] (start, count)
    long NC, int nch, int da, int db, int dg, int do_) {
  __shared__ float lds[kGroupsPerBlock][3 * kMaxDim];
  int group = threadIdx.x / kGroup;
  int lane = threadIdx.x % kGroup;
  long i = (long)blockIdx.x * kGroupsPerBlock + group;
  if (i >= NC) return;
  long e = i / nch;

  float* la = lds[group];
  float* lb = la + kMaxDim;
  float* lc = lb + kMaxDim;
  // cooperative stage of the three rows
  for (int k = lane; k < da; k += kGroup) la[k] = to_f32(A[i * da + k]);
  for (int k = lane; k < db; k += kGroup) lb[k] = to_f32(B[e * db + k]);
  for (int k = lane; k < dg; k += kGroup) lc[k] = to_f32(C[i * dg + k]);
  __builtin_amdgcn_s_waitcnt(0);  // lgkmcnt(0): LDS writes visible
  __builtin_amdgcn_wave_barrier();

  if (lane < do_) {
    int2 r = o_ranges[lane];
    float acc = 0.f;
    for (int k = r.x; k < r.x + r.y; ++k) {
      int4 q = entries[k];
      acc += coefs[k] * la[q.x] * lb[q.y] * lc[q.z];
    }
    out[i * do_ + lane] = from_f32<T>(acc);
  }
}

// out[e,b] = sum_c sum over entries(coef, a, b, g, o) of
//            coef * A[e,c,a] * C[e,c,g] * D[e,c,o]
// (the B-slot gradient: reduce over channels).
template <typename T>
__global__ void etp_reduce_kernel(
    const T* __restrict__ A, const T* __restrict__ C,
    const T* __restrict__ D, float* __restrict__ out,
    const int4* __restrict__ entries,   // (a, b, g, o) any order
    const float* __restrict__ coefs, int n_ent,
    long E, int nch, int da, int db, int dg, int do_) {
  // one wave (64 lanes) per edge, lane strides channels
  __shared__ float lds[4][64 * 12];  // per-wave per-lane db-acc (db<=12)
  int wave = threadIdx.x / 64;
  int lane = threadIdx.x % 64;
  long e = (long)blockIdx.x * 4 + wave;
  bool active = e < E;
  float* my = &lds[wave][lane * db];
  for (int b = 0; b < db; ++b) my[b] = 0.f;
  for (int c = lane; active && c < nch; c += 64) {
    long i = e * nch + c;
    const T* a = A + i * da;
    const T* cc = C + i * dg;
    const T* dd = D + i * do_;
    for (int k = 0; k < n_ent; ++k) {
      int4 q = entries[k];
      my[q.y] += coefs[k] * to_f32(a[q.x]) * to_f32(cc[q.z])
                 * to_f32(dd[q.w]);
    }
  }
  __syncthreads();
  // tree-reduce the 64 per-lane slices
  for (int off = 32; off >= 1; off >>= 1) {
    if (lane < off) {
      float* other = &lds[wave][(lane + off) * db];
      for (int b = 0; b < db; ++b) my[b] += other[b];
    }
    __syncthreads();
  }
  if (active && lane == 0) {
    for (int b = 0; b < db; ++b) out[e * db + b] = my[b];
  }
}

}  // namespace

static hipStream_t etp_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

torch::Tensor etp_general(torch::Tensor A, torch::Tensor B, torch::Tensor C,
                          torch::Tensor entries, torch::Tensor coefs,
                          torch::Tensor o_ranges, long do_) {
  TORCH_CHECK(A.is_cuda() && A.is_contiguous());
  TORCH_CHECK(B.is_contiguous() && C.is_contiguous());
  long NC = A.size(0) * A.size(1);
  int nch = A.size(1);
  int da = A.size(2), db = B.size(1), dg = C.size(2);
  TORCH_CHECK(da <= 40 && db <= 40 && dg <= 40 && do_ <= 32,
              "etp dims exceed kernel limits");
  auto out = torch::empty({A.size(0), A.size(1), do_}, A.options());
  if (NC == 0) return out;
  long blocks = (NC + kGroupsPerBlock - 1) / kGroupsPerBlock;
  // fp64 is routed to the eager path in Python (float LDS staging here)
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, A.scalar_type(),
      "etp_general", [&] {
        hipLaunchKernelGGL(
            etp_general_kernel<scalar_t>, dim3(blocks),
            dim3(kGroup * kGroupsPerBlock), 0, etp_stream(),
            A.data_ptr<scalar_t>(), B.data_ptr<scalar_t>(),
            C.data_ptr<scalar_t>(), out.data_ptr<scalar_t>(),
            reinterpret_cast<const int4*>(entries.data_ptr<int>()),
            coefs.data_ptr<float>(),
            reinterpret_cast<const int2*>(o_ranges.data_ptr<int>()),
            NC, nch, da, db, dg, (int)do_);
      });
  return out;
}

torch::Tensor etp_reduce(torch::Tensor A, torch::Tensor C, torch::Tensor D,
                         torch::Tensor entries, torch::Tensor coefs,
                         long db) {
  TORCH_CHECK(A.is_cuda() && A.is_contiguous());
  long E = A.size(0);
  int nch = A.size(1);
  int da = A.size(2), dg = C.size(2), do_ = D.size(2);
  TORCH_CHECK(db <= 12, "etp_reduce db limit");
  auto out = torch::zeros({E, db}, A.options().dtype(torch::kFloat));
  if (E == 0) return out.to(A.scalar_type());
  long blocks = (E + 3) / 4;
  int n_ent = entries.size(0);
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, A.scalar_type(),
      "etp_reduce", [&] {
        hipLaunchKernelGGL(
            etp_reduce_kernel<scalar_t>, dim3(blocks), dim3(256), 0,
            etp_stream(), A.data_ptr<scalar_t>(), C.data_ptr<scalar_t>(),
            D.data_ptr<scalar_t>(), out.data_ptr<float>(),
            reinterpret_cast<const int4*>(entries.data_ptr<int>()),
            coefs.data_ptr<float>(), n_ent, E, nch, da, (int)db, dg, do_);
      });
  return out.to(A.scalar_type());
}
