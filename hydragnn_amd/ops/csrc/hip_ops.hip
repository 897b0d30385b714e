// hydragnn_amd HIP kernels for MI355X (gfx950, CDNA4).
//
// Segment reductions (gather / scatter) are THE hot aggregation of every
// message-passing stack. Design notes:
//  - wave = 64 lanes; block sizes are multiples of 64.
//  - memory-bound ops: grid-stride loops, coalesced along the feature
//    dim, vectorized where dtype/shape allow.
//  - bf16/f16 scatter accumulates in fp32 (atomicAdd on packed halves is
//    both slow and lossy), cast once at the end.
//  - grid capped at ~2048 blocks with grid-stride (guide G11).
//
// Reference behavior being reimplemented (not copied):
//   torch_scatter.scatter / index_add_ call sites listed in SURVEY.md §2c.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define CHECK_CUDA(x) TORCH_CHECK(x.is_cuda(), #x " must be a GPU tensor")
#define CHECK_CONTIG(x) TORCH_CHECK(x.is_contiguous(), #x " must be contiguous")

namespace {

constexpr int kBlock = 256;

inline int n_blocks(long total, int block, int cap = 2048) {
  long b = (total + block - 1) / block;
  return (int)std::min<long>(b, cap);
}

// -------------------------------------------------------------------------
// gather: out[e, f] = src[index[e], f]
// -------------------------------------------------------------------------
template <typename T>
__global__ void gather_kernel(const T* __restrict__ src,
                              const long* __restrict__ index,
                              T* __restrict__ out, long E, long F) {
  long total = E * F;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    long e = i / F;
    long f = i - e * F;
    out[i] = src[index[e] * F + f];
  }
}

// float4-vectorized variant for F % 4 == 0 (16B/lane coalescing).
template <typename V>
__global__ void gather_kernel_vec(const V* __restrict__ src,
                                  const long* __restrict__ index,
                                  V* __restrict__ out, long E, long Fv) {
  long total = E * Fv;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    long e = i / Fv;
    long f = i - e * Fv;
    out[i] = src[index[e] * Fv + f];
  }
}

// -------------------------------------------------------------------------
// scatter-add (atomic path; CSR segment path below is used when rowptr
// is available).  fp32/fp64 only — half types go through an fp32 buffer.
// -------------------------------------------------------------------------
template <typename T>
__global__ void scatter_add_kernel(const T* __restrict__ src,
                                   const long* __restrict__ index,
                                   T* __restrict__ out, long E, long F) {
  long total = E * F;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    long e = i / F;
    long f = i - e * F;
    atomicAdd(&out[index[e] * F + f], src[i]);
  }
}

__global__ void scatter_add_bf16_kernel(const __hip_bfloat16* __restrict__ src,
                                        const long* __restrict__ index,
                                        float* __restrict__ out, long E,
                                        long F) {
  long total = E * F;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    long e = i / F;
    long f = i - e * F;
    atomicAdd(&out[index[e] * F + f], __bfloat162float(src[i]));
  }
}

__global__ void scatter_add_f16_kernel(const __half* __restrict__ src,
                                       const long* __restrict__ index,
                                       float* __restrict__ out, long E,
                                       long F) {
  long total = E * F;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    long e = i / F;
    long f = i - e * F;
    atomicAdd(&out[index[e] * F + f], __half2float(src[i]));
  }
}

// CSR segment sum over dst-sorted edges: deterministic and
// contention-free (vs the atomic path).  Thread per (row, feature);
// consecutive edges of a row are contiguous src rows -> coalesced
// within each step of the edge loop.
template <typename T, typename ACC>
__global__ void segment_sum_csr_kernel(const T* __restrict__ src,
                                       const long* __restrict__ rowptr,
                                       T* __restrict__ out, long N,
                                       long F) {
  long total = N * F;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    long n = i / F;
    long f = i - n * F;
    ACC acc = (ACC)0;
    long lo = rowptr[n], hi = rowptr[n + 1];
    for (long e = lo; e < hi; ++e) acc += (ACC)src[e * F + f];
    out[i] = (T)acc;
  }
}

template <>
__global__ void segment_sum_csr_kernel<__hip_bfloat16, float>(
    const __hip_bfloat16* __restrict__ src,
    const long* __restrict__ rowptr, __hip_bfloat16* __restrict__ out,
    long N, long F) {
  long total = N * F;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    long n = i / F;
    long f = i - n * F;
    float acc = 0.f;
    long lo = rowptr[n], hi = rowptr[n + 1];
    for (long e = lo; e < hi; ++e)
      acc += __bfloat162float(src[e * F + f]);
    out[i] = __float2bfloat16(acc);
  }
}

// Indexed CSR segment sum: out[r] = sum_{e in row r} src[perm[e]].
// Used as gather's backward when the caller precomputed an index-sort
// (scatter-by-unsorted-index without atomics: deterministic, no L2
// contention on hot rows).
template <typename T, typename ACC>
__global__ void segment_sum_csr_idx_kernel(
    const T* __restrict__ src, const long* __restrict__ rowptr,
    const long* __restrict__ perm, T* __restrict__ out, long N,
    long F) {
  long total = N * F;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    long n = i / F;
    long f = i - n * F;
    ACC acc = (ACC)0;
    long lo = rowptr[n], hi = rowptr[n + 1];
    for (long e = lo; e < hi; ++e) acc += (ACC)src[perm[e] * F + f];
    out[i] = (T)acc;
  }
}

template <>
__global__ void segment_sum_csr_idx_kernel<__hip_bfloat16, float>(
    const __hip_bfloat16* __restrict__ src,
    const long* __restrict__ rowptr, const long* __restrict__ perm,
    __hip_bfloat16* __restrict__ out, long N, long F) {
  long total = N * F;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    long n = i / F;
    long f = i - n * F;
    float acc = 0.f;
    long lo = rowptr[n], hi = rowptr[n + 1];
    for (long e = lo; e < hi; ++e)
      acc += __bfloat162float(src[perm[e] * F + f]);
    out[i] = __float2bfloat16(acc);
  }
}

__global__ void count_kernel(const long* __restrict__ index,
                             float* __restrict__ count, long E) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < E;
       i += (long)gridDim.x * blockDim.x) {
    atomicAdd(&count[index[i]], 1.0f);
  }
}

template <typename T>
__global__ void divide_rows_kernel(T* __restrict__ out,
                                   const float* __restrict__ count, long N,
                                   long F) {
  long total = N * F;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    long n = i / F;
    float c = count[n];
    if (c > 0.5f) out[i] = out[i] / (T)c;
  }
}

// -------------------------------------------------------------------------
// scatter min/max with argext — two-pass:
//   pass 1: atomic extreme on float-ordered bits
//   pass 2: first (lowest e) matching edge wins the arg slot (atomicMin)
// -------------------------------------------------------------------------
__device__ inline unsigned int float_flip(float f) {
  unsigned int u = __float_as_uint(f);
  return (u & 0x80000000u) ? ~u : (u | 0x80000000u);  // order-preserving
}
__device__ inline float float_unflip(unsigned int u) {
  return __uint_as_float((u & 0x80000000u) ? (u & 0x7fffffffu) : ~u);
}

__global__ void scatter_max_pass1(const float* __restrict__ src,
                                  const long* __restrict__ index,
                                  unsigned int* __restrict__ out_bits, long E,
                                  long F, bool is_max) {
  long total = E * F;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    long e = i / F;
    long f = i - e * F;
    unsigned int bits = float_flip(src[i]);
    unsigned int* slot = &out_bits[index[e] * F + f];
    if (is_max)
      atomicMax(slot, bits);
    else
      atomicMin(slot, bits);
  }
}

__global__ void scatter_max_pass2(const float* __restrict__ src,
                                  const long* __restrict__ index,
                                  const unsigned int* __restrict__ out_bits,
                                  unsigned long long* __restrict__ arg, long E,
                                  long F) {
  long total = E * F;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    long e = i / F;
    long f = i - e * F;
    long slot = index[e] * F + f;
    if (float_flip(src[i]) == out_bits[slot]) {
      atomicMin(&arg[slot], (unsigned long long)e);
    }
  }
}

// ---- fp64 variants (order-preserving flip on 64-bit patterns) ----
__device__ inline unsigned long long double_flip(double f) {
  unsigned long long u = __double_as_longlong(f);
  return (u & 0x8000000000000000ull) ? ~u : (u | 0x8000000000000000ull);
}
__device__ inline double double_unflip(unsigned long long u) {
  return __longlong_as_double(
      (u & 0x8000000000000000ull) ? (u & 0x7fffffffffffffffull) : ~u);
}

__global__ void scatter_max_pass1_f64(const double* __restrict__ src,
                                      const long* __restrict__ index,
                                      unsigned long long* __restrict__ out_bits,
                                      long E, long F, bool is_max) {
  long total = E * F;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    long e = i / F;
    long f = i - e * F;
    unsigned long long bits = double_flip(src[i]);
    unsigned long long* slot = &out_bits[index[e] * F + f];
    if (is_max)
      atomicMax(slot, bits);
    else
      atomicMin(slot, bits);
  }
}

__global__ void scatter_max_pass2_f64(const double* __restrict__ src,
                                      const long* __restrict__ index,
                                      const unsigned long long* __restrict__ out_bits,
                                      unsigned long long* __restrict__ arg,
                                      long E, long F) {
  long total = E * F;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    long e = i / F;
    long f = i - e * F;
    long slot = index[e] * F + f;
    if (double_flip(src[i]) == out_bits[slot]) {
      atomicMin(&arg[slot], (unsigned long long)e);
    }
  }
}

__global__ void scatter_max_finalize_f64(
    const unsigned long long* __restrict__ bits,
    const unsigned long long* __restrict__ arg, double* __restrict__ out,
    long* __restrict__ arg_out, long total, bool is_max) {
  unsigned long long empty =
      is_max ? double_flip(-INFINITY) : double_flip(INFINITY);
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    bool present = bits[i] != empty;
    out[i] = present ? double_unflip(bits[i]) : 0.0;
    arg_out[i] = present ? (long)arg[i] : -1;
  }
}

__global__ void scatter_max_finalize(const unsigned int* __restrict__ bits,
                                     const unsigned long long* __restrict__ arg,
                                     float* __restrict__ out,
                                     long* __restrict__ arg_out, long total,
                                     bool is_max) {
  unsigned int empty =
      is_max ? float_flip(-INFINITY) : float_flip(INFINITY);
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    bool present = bits[i] != empty;
    out[i] = present ? float_unflip(bits[i]) : 0.0f;
    arg_out[i] = present ? (long)arg[i] : -1;
  }
}

// -------------------------------------------------------------------------
// radius graph (brute force within graph, count+fill; capping by
// distance order is finished in Python with device torch ops)
// -------------------------------------------------------------------------
__global__ void radius_count_kernel(const float* __restrict__ pos,
                                    const long* __restrict__ graph_of,
                                    const long* __restrict__ gptr, long N,
                                    float r2, bool loop,
                                    int* __restrict__ count) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < N;
       i += (long)gridDim.x * blockDim.x) {
    long g = graph_of[i];
    long lo = gptr[g], hi = gptr[g + 1];
    float xi = pos[i * 3], yi = pos[i * 3 + 1], zi = pos[i * 3 + 2];
    int c = 0;
    for (long j = lo; j < hi; ++j) {
      if (!loop && j == i) continue;
      float dx = pos[j * 3] - xi, dy = pos[j * 3 + 1] - yi,
            dz = pos[j * 3 + 2] - zi;
      if (dx * dx + dy * dy + dz * dz <= r2) ++c;
    }
    count[i] = c;
  }
}

__global__ void radius_fill_kernel(const float* __restrict__ pos,
                                   const long* __restrict__ graph_of,
                                   const long* __restrict__ gptr,
                                   const long* __restrict__ offs, long N,
                                   float r2, bool loop,
                                   long* __restrict__ src_out,
                                   long* __restrict__ dst_out,
                                   float* __restrict__ dist_out) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < N;
       i += (long)gridDim.x * blockDim.x) {
    long g = graph_of[i];
    long lo = gptr[g], hi = gptr[g + 1];
    float xi = pos[i * 3], yi = pos[i * 3 + 1], zi = pos[i * 3 + 2];
    long w = offs[i];
    for (long j = lo; j < hi; ++j) {
      if (!loop && j == i) continue;
      float dx = pos[j * 3] - xi, dy = pos[j * 3 + 1] - yi,
            dz = pos[j * 3 + 2] - zi;
      float d2 = dx * dx + dy * dy + dz * dz;
      if (d2 <= r2) {
        src_out[w] = j;   // src = neighbor
        dst_out[w] = i;   // dst = center
        dist_out[w] = sqrtf(d2);
        ++w;
      }
    }
  }
}

}  // namespace

// ===========================================================================
// C++ entry points
// ===========================================================================

static hipStream_t cur_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

torch::Tensor gather_fwd(torch::Tensor src, torch::Tensor index) {
  CHECK_CUDA(src); CHECK_CONTIG(src); CHECK_CUDA(index);
  TORCH_CHECK(index.dtype() == torch::kLong);
  long E = index.numel();
  long F = src.numel() / std::max<long>(src.size(0), 1);
  auto sizes = src.sizes().vec();
  sizes[0] = E;
  auto out = torch::empty(sizes, src.options());
  if (E == 0) return out;
  auto idx = index.contiguous();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, src.scalar_type(),
      "gather_fwd", [&] {
        long bytes = F * sizeof(scalar_t);
        if (bytes % 16 == 0) {
          long Fv = bytes / 16;
          hipLaunchKernelGGL(gather_kernel_vec<float4>,
                             dim3(n_blocks(E * Fv, kBlock)), dim3(kBlock), 0,
                             cur_stream(),
                             reinterpret_cast<const float4*>(src.data_ptr<scalar_t>()),
                             idx.data_ptr<long>(),
                             reinterpret_cast<float4*>(out.data_ptr<scalar_t>()),
                             E, Fv);
        } else {
          hipLaunchKernelGGL(gather_kernel<scalar_t>,
                             dim3(n_blocks(E * F, kBlock)), dim3(kBlock), 0,
                             cur_stream(), src.data_ptr<scalar_t>(),
                             idx.data_ptr<long>(), out.data_ptr<scalar_t>(), E,
                             F);
        }
      });
  return out;
}

torch::Tensor scatter_sum_fwd(torch::Tensor src, torch::Tensor index,
                              long dim_size) {
  CHECK_CUDA(src); CHECK_CONTIG(src); CHECK_CUDA(index);
  long E = index.numel();
  long F = E > 0 ? src.numel() / src.size(0) : 1;
  auto sizes = src.sizes().vec();
  sizes[0] = dim_size;
  auto idx = index.contiguous();
  if (src.scalar_type() == at::ScalarType::BFloat16 ||
      src.scalar_type() == at::ScalarType::Half) {
    auto acc = torch::zeros(sizes, src.options().dtype(torch::kFloat));
    if (E > 0) {
      if (src.scalar_type() == at::ScalarType::BFloat16) {
        hipLaunchKernelGGL(scatter_add_bf16_kernel,
                           dim3(n_blocks(E * F, kBlock)), dim3(kBlock), 0,
                           cur_stream(),
                           reinterpret_cast<const __hip_bfloat16*>(src.data_ptr()),
                           idx.data_ptr<long>(), acc.data_ptr<float>(), E, F);
      } else {
        hipLaunchKernelGGL(scatter_add_f16_kernel,
                           dim3(n_blocks(E * F, kBlock)), dim3(kBlock), 0,
                           cur_stream(),
                           reinterpret_cast<const __half*>(src.data_ptr()),
                           idx.data_ptr<long>(), acc.data_ptr<float>(), E, F);
      }
    }
    return acc.to(src.scalar_type());
  }
  auto out = torch::zeros(sizes, src.options());
  if (E == 0) return out;
  AT_DISPATCH_FLOATING_TYPES(src.scalar_type(), "scatter_sum_fwd", [&] {
    hipLaunchKernelGGL(scatter_add_kernel<scalar_t>,
                       dim3(n_blocks(E * F, kBlock)), dim3(kBlock), 0,
                       cur_stream(), src.data_ptr<scalar_t>(),
                       idx.data_ptr<long>(), out.data_ptr<scalar_t>(), E, F);
  });
  return out;
}

torch::Tensor segment_sum_csr(torch::Tensor src, torch::Tensor rowptr,
                              c10::optional<torch::Tensor> perm) {
  CHECK_CUDA(src); CHECK_CONTIG(src);
  long N = rowptr.numel() - 1;
  long F = src.numel() / std::max<long>(src.size(0), 1);
  auto sizes = src.sizes().vec();
  sizes[0] = N;
  auto out = torch::empty(sizes, src.options());
  auto rp = rowptr.contiguous();
  if (N == 0) return out;
  torch::Tensor pc;
  const long* perm_ptr = nullptr;
  if (perm.has_value()) {
    pc = perm->contiguous();
    perm_ptr = pc.data_ptr<long>();
  }
  if (src.scalar_type() == at::ScalarType::BFloat16) {
    if (perm_ptr) {
      hipLaunchKernelGGL(
          (segment_sum_csr_idx_kernel<__hip_bfloat16, float>),
          dim3(n_blocks(N * F, kBlock, 8192)), dim3(kBlock), 0,
          cur_stream(),
          reinterpret_cast<const __hip_bfloat16*>(src.data_ptr()),
          rp.data_ptr<long>(), perm_ptr,
          reinterpret_cast<__hip_bfloat16*>(out.data_ptr()), N, F);
    } else {
      hipLaunchKernelGGL(
          (segment_sum_csr_kernel<__hip_bfloat16, float>),
          dim3(n_blocks(N * F, kBlock, 8192)), dim3(kBlock), 0,
          cur_stream(),
          reinterpret_cast<const __hip_bfloat16*>(src.data_ptr()),
          rp.data_ptr<long>(),
          reinterpret_cast<__hip_bfloat16*>(out.data_ptr()), N, F);
    }
    return out;
  }
  AT_DISPATCH_FLOATING_TYPES_AND(at::ScalarType::Half, src.scalar_type(),
                                 "segment_sum_csr", [&] {
    if (perm_ptr) {
      hipLaunchKernelGGL((segment_sum_csr_idx_kernel<scalar_t, scalar_t>),
                         dim3(n_blocks(N * F, kBlock, 8192)), dim3(kBlock),
                         0, cur_stream(), src.data_ptr<scalar_t>(),
                         rp.data_ptr<long>(), perm_ptr,
                         out.data_ptr<scalar_t>(), N, F);
    } else {
      hipLaunchKernelGGL((segment_sum_csr_kernel<scalar_t, scalar_t>),
                         dim3(n_blocks(N * F, kBlock, 8192)), dim3(kBlock),
                         0, cur_stream(), src.data_ptr<scalar_t>(),
                         rp.data_ptr<long>(), out.data_ptr<scalar_t>(), N,
                         F);
    }
  });
  return out;
}

std::vector<torch::Tensor> scatter_mean_fwd(torch::Tensor src,
                                            torch::Tensor index,
                                            long dim_size) {
  auto out = scatter_sum_fwd(src, index, dim_size);
  long E = index.numel();
  long F = E > 0 ? src.numel() / src.size(0) : 1;
  auto count = torch::zeros({dim_size}, src.options().dtype(torch::kFloat));
  auto idx = index.contiguous();
  if (E > 0) {
    hipLaunchKernelGGL(count_kernel, dim3(n_blocks(E, kBlock)), dim3(kBlock),
                       0, cur_stream(), idx.data_ptr<long>(),
                       count.data_ptr<float>(), E);
    AT_DISPATCH_FLOATING_TYPES_AND2(
        at::ScalarType::BFloat16, at::ScalarType::Half, out.scalar_type(),
        "divide_rows", [&] {
          hipLaunchKernelGGL(divide_rows_kernel<scalar_t>,
                             dim3(n_blocks(dim_size * F, kBlock)),
                             dim3(kBlock), 0, cur_stream(),
                             out.data_ptr<scalar_t>(), count.data_ptr<float>(),
                             dim_size, F);
        });
  }
  return {out, count};
}

std::vector<torch::Tensor> scatter_minmax_fwd_f64(torch::Tensor src,
                                                  torch::Tensor index,
                                                  long dim_size,
                                                  bool is_max) {
  auto srcd = src.contiguous();
  long E = index.numel();
  long F = E > 0 ? srcd.numel() / srcd.size(0) : 1;
  auto sizes = src.sizes().vec();
  sizes[0] = dim_size;
  long total = dim_size * F;
  auto idx = index.contiguous();
  auto bits = torch::empty(sizes, src.options().dtype(torch::kLong));
  {
    double fill = is_max ? -INFINITY : INFINITY;
    unsigned long long raw;
    memcpy(&raw, &fill, 8);
    unsigned long long u = (raw & 0x8000000000000000ull) ? ~raw
                           : (raw | 0x8000000000000000ull);
    bits.fill_((long)u);
  }
  auto arg64 = torch::full(sizes, (long)0x7fffffffffffffffLL,
                           src.options().dtype(torch::kLong));
  auto out = torch::empty(sizes, src.options());
  auto arg = torch::empty(sizes, src.options().dtype(torch::kLong));
  if (E > 0) {
    hipLaunchKernelGGL(scatter_max_pass1_f64, dim3(n_blocks(E * F, kBlock)),
                       dim3(kBlock), 0, cur_stream(), srcd.data_ptr<double>(),
                       idx.data_ptr<long>(),
                       reinterpret_cast<unsigned long long*>(bits.data_ptr<long>()),
                       E, F, is_max);
    hipLaunchKernelGGL(scatter_max_pass2_f64, dim3(n_blocks(E * F, kBlock)),
                       dim3(kBlock), 0, cur_stream(), srcd.data_ptr<double>(),
                       idx.data_ptr<long>(),
                       reinterpret_cast<unsigned long long*>(bits.data_ptr<long>()),
                       reinterpret_cast<unsigned long long*>(arg64.data_ptr<long>()),
                       E, F);
  }
  hipLaunchKernelGGL(scatter_max_finalize_f64,
                     dim3(n_blocks(total, kBlock)), dim3(kBlock), 0,
                     cur_stream(),
                     reinterpret_cast<unsigned long long*>(bits.data_ptr<long>()),
                     reinterpret_cast<unsigned long long*>(arg64.data_ptr<long>()),
                     out.data_ptr<double>(), arg.data_ptr<long>(), total,
                     is_max);
  return {out, arg};
}

std::vector<torch::Tensor> scatter_minmax_fwd(torch::Tensor src,
                                              torch::Tensor index,
                                              long dim_size, bool is_max) {
  CHECK_CUDA(src); CHECK_CUDA(index);
  if (src.scalar_type() == at::ScalarType::Double) {
    return scatter_minmax_fwd_f64(src, index, dim_size, is_max);
  }
  auto srcf = src.contiguous().to(torch::kFloat);
  long E = index.numel();
  long F = E > 0 ? srcf.numel() / srcf.size(0) : 1;
  auto sizes = src.sizes().vec();
  sizes[0] = dim_size;
  long total = dim_size * F;
  auto idx = index.contiguous();
  auto bits = torch::empty(sizes, src.options().dtype(torch::kInt));
  {
    float fill = is_max ? -INFINITY : INFINITY;
    unsigned int u;
    // host-side flip of the fill value
    unsigned int raw;
    memcpy(&raw, &fill, 4);
    u = (raw & 0x80000000u) ? ~raw : (raw | 0x80000000u);
    bits.fill_((int)u);
  }
  auto arg64 = torch::full(sizes, (long)0x7fffffffffffffffLL,
                           src.options().dtype(torch::kLong));
  auto out = torch::empty(sizes, src.options().dtype(torch::kFloat));
  auto arg = torch::empty(sizes, src.options().dtype(torch::kLong));
  if (E > 0) {
    hipLaunchKernelGGL(scatter_max_pass1, dim3(n_blocks(E * F, kBlock)),
                       dim3(kBlock), 0, cur_stream(), srcf.data_ptr<float>(),
                       idx.data_ptr<long>(),
                       reinterpret_cast<unsigned int*>(bits.data_ptr<int>()),
                       E, F, is_max);
    hipLaunchKernelGGL(scatter_max_pass2, dim3(n_blocks(E * F, kBlock)),
                       dim3(kBlock), 0, cur_stream(), srcf.data_ptr<float>(),
                       idx.data_ptr<long>(),
                       reinterpret_cast<unsigned int*>(bits.data_ptr<int>()),
                       reinterpret_cast<unsigned long long*>(arg64.data_ptr<long>()),
                       E, F);
  }
  hipLaunchKernelGGL(scatter_max_finalize, dim3(n_blocks(total, kBlock)),
                     dim3(kBlock), 0, cur_stream(),
                     reinterpret_cast<unsigned int*>(bits.data_ptr<int>()),
                     reinterpret_cast<unsigned long long*>(arg64.data_ptr<long>()),
                     out.data_ptr<float>(), arg.data_ptr<long>(), total,
                     is_max);
  return {out.to(src.scalar_type()), arg};
}

std::vector<torch::Tensor> radius_pairs(torch::Tensor pos, torch::Tensor batch,
                                        torch::Tensor gptr, double r,
                                        bool loop) {
  CHECK_CUDA(pos); CHECK_CONTIG(pos);
  long N = pos.size(0);
  float r2 = (float)(r * r);
  auto count = torch::zeros({N}, pos.options().dtype(torch::kInt));
  auto b = batch.contiguous();
  auto gp = gptr.contiguous();
  hipLaunchKernelGGL(radius_count_kernel, dim3(n_blocks(N, kBlock)),
                     dim3(kBlock), 0, cur_stream(), pos.data_ptr<float>(),
                     b.data_ptr<long>(), gp.data_ptr<long>(), N, r2, loop,
                     count.data_ptr<int>());
  auto offs = torch::zeros({N}, pos.options().dtype(torch::kLong));
  auto csum = count.to(torch::kLong).cumsum(0);
  offs.slice(0, 1, N).copy_(csum.slice(0, 0, N - 1));
  long E = N > 0 ? csum[-1].item<long>() : 0;
  auto src = torch::empty({E}, pos.options().dtype(torch::kLong));
  auto dst = torch::empty({E}, pos.options().dtype(torch::kLong));
  auto dist = torch::empty({E}, pos.options().dtype(torch::kFloat));
  if (E > 0) {
    hipLaunchKernelGGL(radius_fill_kernel, dim3(n_blocks(N, kBlock)),
                       dim3(kBlock), 0, cur_stream(), pos.data_ptr<float>(),
                       b.data_ptr<long>(), gp.data_ptr<long>(),
                       offs.data_ptr<long>(), N, r2, loop,
                       src.data_ptr<long>(), dst.data_ptr<long>(),
                       dist.data_ptr<float>());
  }
  return {src, dst, dist};
}

// defined in mfma_linear.hip
torch::Tensor mfma_linear(torch::Tensor A, torch::Tensor B,
                          c10::optional<torch::Tensor> bias, bool trans_b);

// defined in etp.hip
torch::Tensor etp_general(torch::Tensor A, torch::Tensor B, torch::Tensor C,
                          torch::Tensor entries, torch::Tensor coefs,
                          torch::Tensor o_ranges, long do_,
                          c10::optional<torch::Tensor> ai,
                          c10::optional<torch::Tensor> bi,
                          c10::optional<torch::Tensor> ci, long n_rows);
torch::Tensor etp_nodesum(torch::Tensor A, torch::Tensor B,
                          torch::Tensor C, torch::Tensor entries,
                          torch::Tensor coefs, long do_,
                          torch::Tensor rowptr,
                          c10::optional<torch::Tensor> ai,
                          c10::optional<torch::Tensor> bi,
                          c10::optional<torch::Tensor> ci);
torch::Tensor etp_reduce(torch::Tensor A, torch::Tensor C, torch::Tensor D,
                         torch::Tensor entries, torch::Tensor coefs,
                         long db,
                         c10::optional<torch::Tensor> ai,
                         c10::optional<torch::Tensor> ci,
                         c10::optional<torch::Tensor> di, long n_rows);

// defined in varlen_attn.hip
torch::Tensor varlen_attention(torch::Tensor Q, torch::Tensor K,
                               torch::Tensor V, torch::Tensor ptr);

// defined in irreps_linear.hip
torch::Tensor irreps_linear(torch::Tensor X, torch::Tensor W,
                            torch::Tensor lmap,
                            c10::optional<torch::Tensor> bias,
                            bool trans_w,
                            c10::optional<torch::Tensor> add);
torch::Tensor irreps_linear_gw(torch::Tensor X, torch::Tensor G,
                               torch::Tensor lmap, long L,
                               long nblocks);
// defined in gemv.hip
torch::Tensor gemv_small_n(torch::Tensor A, torch::Tensor W,
                           c10::optional<torch::Tensor> bias);
// defined in fused_adamw.hip
void fused_adamw(torch::Tensor p, torch::Tensor g, torch::Tensor m,
                 torch::Tensor v, torch::Tensor step, double lr,
                 double beta1, double beta2, double eps, double wd);
void fused_adamw_bf16(torch::Tensor p, torch::Tensor g,
                      torch::Tensor master, torch::Tensor m,
                      torch::Tensor v, torch::Tensor step, double lr,
                      double beta1, double beta2, double eps,
                      double wd);
// defined in radius.hip
std::vector<torch::Tensor> radius_pairs_t(torch::Tensor pos,
                                          torch::Tensor batch,
                                          torch::Tensor gptr, double r,
                                          bool loop,
                                          c10::optional<torch::Tensor> shifts);
std::vector<torch::Tensor> radius_pairs_cells(
    torch::Tensor pos, torch::Tensor order, torch::Tensor cell_of,
    torch::Tensor cell_start, long ncx, long ncy, long ncz, double r,
    bool loop);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("etp_general", &etp_general, "fused ETP contraction (HIP)",
        pybind11::arg("A"), pybind11::arg("B"), pybind11::arg("C"),
        pybind11::arg("entries"), pybind11::arg("coefs"),
        pybind11::arg("o_ranges"), pybind11::arg("do_"),
        pybind11::arg("ai") = pybind11::none(),
        pybind11::arg("bi") = pybind11::none(),
        pybind11::arg("ci") = pybind11::none(),
        pybind11::arg("n_rows") = 0);
  m.def("etp_nodesum", &etp_nodesum, "fused gather+TP+segment sum (HIP)",
        pybind11::arg("A"), pybind11::arg("B"), pybind11::arg("C"),
        pybind11::arg("entries"), pybind11::arg("coefs"),
        pybind11::arg("do_"), pybind11::arg("rowptr"),
        pybind11::arg("ai") = pybind11::none(),
        pybind11::arg("bi") = pybind11::none(),
        pybind11::arg("ci") = pybind11::none());
  m.def("mfma_linear", &mfma_linear, "bf16 MFMA linear (HIP)",
        pybind11::arg("A"), pybind11::arg("B"),
        pybind11::arg("bias") = pybind11::none(),
        pybind11::arg("trans_b") = true);
  m.def("etp_reduce", &etp_reduce, "fused ETP channel-reduce (HIP)",
        pybind11::arg("A"), pybind11::arg("C"), pybind11::arg("D"),
        pybind11::arg("entries"), pybind11::arg("coefs"),
        pybind11::arg("db"),
        pybind11::arg("ai") = pybind11::none(),
        pybind11::arg("ci") = pybind11::none(),
        pybind11::arg("di") = pybind11::none(),
        pybind11::arg("n_rows") = 0);
  m.def("gather_fwd", &gather_fwd, "gather rows (HIP)");
  m.def("scatter_sum_fwd", &scatter_sum_fwd, "scatter-add (HIP)");
  m.def("segment_sum_csr", &segment_sum_csr, "CSR segment sum (HIP)",
        pybind11::arg("src"), pybind11::arg("rowptr"),
        pybind11::arg("perm") = pybind11::none());
  m.def("scatter_mean_fwd", &scatter_mean_fwd, "scatter-mean (HIP)");
  m.def("scatter_minmax_fwd", &scatter_minmax_fwd, "scatter-min/max (HIP)");
  m.def("radius_pairs", &radius_pairs, "radius pair enumeration (HIP)");
  m.def("fused_adamw", &fused_adamw,
        "single-kernel flat AdamW (HIP)");
  m.def("fused_adamw_bf16", &fused_adamw_bf16,
        "single-kernel flat AdamW, bf16 params + fp32 master (HIP)");
  m.def("radius_pairs_cells", &radius_pairs_cells,
        "cell-list radius pairs for large graphs (HIP)");
  m.def("radius_pairs_t", &radius_pairs_t,
        "tiled fp32/fp64 radius pairs, open or periodic (HIP)",
        pybind11::arg("pos"), pybind11::arg("batch"),
        pybind11::arg("gptr"), pybind11::arg("r"),
        pybind11::arg("loop") = false,
        pybind11::arg("shifts") = pybind11::none());
  m.def("varlen_attention", &varlen_attention,
        "segment-varlen attention (HIP)");
  m.def("irreps_linear", &irreps_linear,
        "per-l channel-mixing MFMA linear (HIP)",
        pybind11::arg("X"), pybind11::arg("W"), pybind11::arg("lmap"),
        pybind11::arg("bias") = pybind11::none(),
        pybind11::arg("trans_w") = false,
        pybind11::arg("add") = pybind11::none());
  m.def("gemv_small_n", &gemv_small_n,
        "narrow-output bf16 linear (HIP)",
        pybind11::arg("A"), pybind11::arg("W"),
        pybind11::arg("bias") = pybind11::none());
  m.def("irreps_linear_gw", &irreps_linear_gw,
        "irreps-linear weight-grad partials (HIP)",
        pybind11::arg("X"), pybind11::arg("G"), pybind11::arg("lmap"),
        pybind11::arg("L"), pybind11::arg("nblocks") = 512);
}
