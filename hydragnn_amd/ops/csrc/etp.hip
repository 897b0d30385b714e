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

// accumulator type: fp32 for bf16/half/float inputs, fp64 for double
// (fp64 ETP support: slices and channel reductions keep full width)
template <typename T> struct acc_of { using type = float; };
template <> struct acc_of<double> { using type = double; };

// out[i,c,o] = sum over entries(coef, a, b, g, o) of
//              coef * A[i,c,a] * B[i,b] * C[i,c,g]
//
// One THREAD per (i, c): its A/B/C rows and do-accumulator live in a
// per-thread LDS slice (runtime entry indices would force register
// arrays to scratch otherwise).  Every thread walks the SAME entry
// list in lockstep (entry words are wave-uniform LDS broadcasts), so
// there is no divergence; global loads/stores are per-thread
// contiguous rows -> coalesced across adjacent threads.  Slice stride
// is padded to an odd word count to spread LDS banks.
template <typename T>
__global__ void etp_general_kernel(
    const T* __restrict__ A, const T* __restrict__ B,
    const T* __restrict__ C, T* __restrict__ out,
    const int4* __restrict__ entries,
    const float* __restrict__ coefs, int n_ent,
    long NC, int nch, int da, int db, int dg, int do_,
    const long* __restrict__ ai, const long* __restrict__ bi,
    const long* __restrict__ ci) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  using ACC = typename acc_of<T>::type;
  const int stride = ((da + db + dg + do_) | 1);  // odd word stride
  ACC* slices = reinterpret_cast<ACC*>(smem);
  int4* ent_lds = reinterpret_cast<int4*>(
      smem + (size_t)blockDim.x * stride * sizeof(ACC));
  float* coef_lds = reinterpret_cast<float*>(ent_lds + n_ent);

  // stage the entry table once per block
  for (int k = threadIdx.x; k < n_ent; k += blockDim.x) {
    ent_lds[k] = entries[k];
    coef_lds[k] = coefs[k];
  }

  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  ACC* my = slices + (size_t)threadIdx.x * stride;
  ACC* ma = my;
  ACC* mb = ma + da;
  ACC* mc = mb + db;
  ACC* mo = mc + dg;
  if (i < NC) {
    long e = i / nch;
    int c = (int)(i - e * nch);
    long ea = ai ? ai[e] : e;
    long eb = bi ? bi[e] : e;
    long ec = ci ? ci[e] : e;
    const T* ap = A + (ea * nch + c) * da;
    const T* bp = B + eb * db;
    const T* cp = C + (ec * nch + c) * dg;
    for (int k = 0; k < da; ++k) ma[k] = (ACC)ap[k];
    for (int k = 0; k < db; ++k) mb[k] = (ACC)bp[k];
    for (int k = 0; k < dg; ++k) mc[k] = (ACC)cp[k];
    for (int k = 0; k < do_; ++k) mo[k] = 0.f;
  }
  __syncthreads();
  if (i < NC) {
    for (int k = 0; k < n_ent; ++k) {
      int4 q = ent_lds[k];
      mo[q.w] += coef_lds[k] * ma[q.x] * mb[q.y] * mc[q.z];
    }
    T* op = out + i * do_;
    for (int k = 0; k < do_; ++k) op[k] = (T)mo[k];
  }
}

// Register-accumulation variant (LDS-pressure fallback): the entry table is
// sorted by output slot with per-output (start, count) ranges
// (ETPTable.device_tensors), so each output accumulates in a REGISTER
// and stores once — the r1 kernel's mo[q.w] += ... formed a serially
// dependent LDS read-modify-write chain (consecutive entries hit the
// same address; PMC: SQ busy only ~13% of wall time = latency-bound).
// The LDS accumulator slice disappears too (smaller stride -> higher
// occupancy).
template <typename T>
__global__ void etp_general_racc_kernel(
    const T* __restrict__ A, const T* __restrict__ B,
    const T* __restrict__ C, T* __restrict__ out,
    const int4* __restrict__ entries,
    const float* __restrict__ coefs,
    const int2* __restrict__ o_ranges, int n_ent,
    long NC, int nch, int da, int db, int dg, int do_,
    const long* __restrict__ ai, const long* __restrict__ bi,
    const long* __restrict__ ci) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  using ACC = typename acc_of<T>::type;
  const int stride = ((da + db + dg) | 1);
  ACC* slices = reinterpret_cast<ACC*>(smem);
  int4* ent_lds = reinterpret_cast<int4*>(
      smem + (size_t)blockDim.x * stride * sizeof(ACC));
  float* coef_lds = reinterpret_cast<float*>(ent_lds + n_ent);
  int2* rng_lds = reinterpret_cast<int2*>(coef_lds + n_ent);

  for (int k = threadIdx.x; k < n_ent; k += blockDim.x) {
    ent_lds[k] = entries[k];
    coef_lds[k] = coefs[k];
  }
  for (int k = threadIdx.x; k < do_; k += blockDim.x)
    rng_lds[k] = o_ranges[k];

  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  ACC* my = slices + (size_t)threadIdx.x * stride;
  ACC* ma = my;
  ACC* mb = ma + da;
  ACC* mc = mb + db;
  if (i < NC) {
    long e = i / nch;
    int c = (int)(i - e * nch);
    long ea = ai ? ai[e] : e;
    long eb = bi ? bi[e] : e;
    long ec = ci ? ci[e] : e;
    const T* ap = A + (ea * nch + c) * da;
    const T* bp = B + eb * db;
    const T* cp = C + (ec * nch + c) * dg;
    for (int k = 0; k < da; ++k) ma[k] = (ACC)ap[k];
    for (int k = 0; k < db; ++k) mb[k] = (ACC)bp[k];
    for (int k = 0; k < dg; ++k) mc[k] = (ACC)cp[k];
  }
  __syncthreads();
  if (i >= NC) return;
  T* op = out + i * do_;
  for (int o = 0; o < do_; ++o) {
    int s = rng_lds[o].x;
    int cnt = rng_lds[o].y;
    ACC acc = (ACC)0;
    for (int k = s; k < s + cnt; ++k) {
      int4 q = ent_lds[k];
      acc += coef_lds[k] * ma[q.x] * mb[q.y] * mc[q.z];
    }
    op[o] = (T)acc;
  }
}

// Occupancy variant of etp_general: A/B/C rows are PRIVATE per thread
// and only ~1-3 cache lines each, so after first touch they are
// L1-hot — staging them in LDS buys nothing but caps the block count
// at 1/CU (zero latency hiding; PMC showed SQ busy ~10% of kernel
// wall time).  Here only the runtime-indexed OUTPUT accumulator lives
// in LDS; operand reads go straight to L1.  The host picks whichever
// variant yields more blocks/CU.
template <typename T>
__global__ void etp_general_l1_kernel(
    const T* __restrict__ A, const T* __restrict__ B,
    const T* __restrict__ C, T* __restrict__ out,
    const int4* __restrict__ entries,
    const float* __restrict__ coefs, int n_ent,
    long NC, int nch, int da, int db, int dg, int do_,
    const long* __restrict__ ai, const long* __restrict__ bi,
    const long* __restrict__ ci) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  using ACC = typename acc_of<T>::type;
  const int stride = do_ | 1;
  ACC* slices = reinterpret_cast<ACC*>(smem);
  int4* ent_lds = reinterpret_cast<int4*>(
      smem + (size_t)blockDim.x * stride * sizeof(ACC));
  float* coef_lds = reinterpret_cast<float*>(ent_lds + n_ent);
  for (int k = threadIdx.x; k < n_ent; k += blockDim.x) {
    ent_lds[k] = entries[k];
    coef_lds[k] = coefs[k];
  }
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  ACC* mo = slices + (size_t)threadIdx.x * stride;
  __syncthreads();
  if (i >= NC) return;
  long e = i / nch;
  int c = (int)(i - e * nch);
  long ea = ai ? ai[e] : e;
  long eb = bi ? bi[e] : e;
  long ec = ci ? ci[e] : e;
  const T* ap = A + (ea * nch + c) * da;
  const T* bp = B + eb * db;
  const T* cp = C + (ec * nch + c) * dg;
  for (int k = 0; k < do_; ++k) mo[k] = 0.f;
  for (int k = 0; k < n_ent; ++k) {
    int4 q = ent_lds[k];
    mo[q.w] += coef_lds[k] * (ACC)ap[q.x] * (ACC)bp[q.y] *
               (ACC)cp[q.z];
  }
  T* op = out + i * do_;
  for (int k = 0; k < do_; ++k) op[k] = (T)mo[k];
}

// out[e,b] = sum_c sum over entries(coef, a, b, g, o) of
//            coef * A[e,c,a] * C[e,c,g] * D[e,c,o]
// (the B-slot gradient: reduce over channels).  Same structure as
// etp_general: one thread per (e, c), rows + db-accumulator in a
// per-thread LDS slice, uniform entry walk; the channel reduction is
// db fp32 atomicAdds per thread (db <= 12, light contention).
template <typename T>
__global__ void etp_reduce_kernel(
    const T* __restrict__ A, const T* __restrict__ C,
    const T* __restrict__ D,
    typename acc_of<T>::type* __restrict__ out,
    const int4* __restrict__ entries,
    const float* __restrict__ coefs, int n_ent,
    long E, int nch, int da, int db, int dg, int do_) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  using ACC = typename acc_of<T>::type;
  const int stride = ((da + dg + do_ + db) | 1);
  ACC* slices = reinterpret_cast<ACC*>(smem);
  int4* ent_lds = reinterpret_cast<int4*>(
      smem + (size_t)blockDim.x * stride * sizeof(ACC));
  float* coef_lds = reinterpret_cast<float*>(ent_lds + n_ent);
  for (int k = threadIdx.x; k < n_ent; k += blockDim.x) {
    ent_lds[k] = entries[k];
    coef_lds[k] = coefs[k];
  }
  long NC = E * nch;
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  ACC* my = slices + (size_t)threadIdx.x * stride;
  ACC* ma = my;
  ACC* mc = ma + da;
  ACC* md = mc + dg;
  ACC* mb = md + do_;
  if (i < NC) {
    const T* ap = A + i * da;
    const T* cp = C + i * dg;
    const T* dp = D + i * do_;
    for (int k = 0; k < da; ++k) ma[k] = (ACC)ap[k];
    for (int k = 0; k < dg; ++k) mc[k] = (ACC)cp[k];
    for (int k = 0; k < do_; ++k) md[k] = (ACC)dp[k];
    for (int k = 0; k < db; ++k) mb[k] = 0.f;
  }
  __syncthreads();
  if (i < NC) {
    for (int k = 0; k < n_ent; ++k) {
      int4 q = ent_lds[k];
      mb[q.y] += coef_lds[k] * ma[q.x] * mc[q.z] * md[q.w];
    }
  }
  long e = i / nch;
  if (nch % 64 == 0) {
    // a wave spans exactly one edge's channels: shuffle-reduce each b
    // across the 64 lanes, then ONE atomic per b per wave
    for (int b = 0; b < db; ++b) {
      ACC v = (i < NC) ? mb[b] : (ACC)0;
      for (int off = 32; off >= 1; off >>= 1)
        v += __shfl_down(v, off, 64);
      if ((threadIdx.x % 64) == 0 && i < NC)
        atomicAdd(&out[e * db + b], v);
    }
  } else if (i < NC) {
    for (int b = 0; b < db; ++b) atomicAdd(&out[e * db + b], mb[b]);
  }
}

// Fused gather + tensor product + segment sum:
//   out[r, c, o] = sum_{e in rowptr[r]..rowptr[r+1]} sum_k coef_k
//                  A[ai[e], c, a] B[bi[e], b] C[ci[e], c, g]
// One thread per (out row, channel); the edge loop restages each
// edge's rows into the thread's LDS slice — removes the materialized
// per-edge message tensor, its scatter pass, and the standalone node
// gather of the unfused pipeline.
template <typename T>
__global__ void etp_nodesum_kernel(
    const T* __restrict__ A, const T* __restrict__ B,
    const T* __restrict__ C, T* __restrict__ out,
    const int4* __restrict__ entries,
    const float* __restrict__ coefs, int n_ent,
    const long* __restrict__ rowptr, long R, int nch,
    int da, int db, int dg, int do_,
    const long* __restrict__ ai, const long* __restrict__ bi,
    const long* __restrict__ ci) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  using ACC = typename acc_of<T>::type;
  const int stride = ((da + db + dg + do_) | 1);
  ACC* slices = reinterpret_cast<ACC*>(smem);
  int4* ent_lds = reinterpret_cast<int4*>(
      smem + (size_t)blockDim.x * stride * sizeof(ACC));
  float* coef_lds = reinterpret_cast<float*>(ent_lds + n_ent);
  for (int k = threadIdx.x; k < n_ent; k += blockDim.x) {
    ent_lds[k] = entries[k];
    coef_lds[k] = coefs[k];
  }
  long RC = R * nch;
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  ACC* my = slices + (size_t)threadIdx.x * stride;
  ACC* ma = my;
  ACC* mb = ma + da;
  ACC* mc = mb + db;
  ACC* mo = mc + dg;
  __syncthreads();
  if (i >= RC) return;
  long r = i / nch;
  int c = (int)(i - r * nch);
  for (int k = 0; k < do_; ++k) mo[k] = 0.f;
  long lo = rowptr[r], hi = rowptr[r + 1];
  for (long e = lo; e < hi; ++e) {
    long ea = ai ? ai[e] : e;
    long eb = bi ? bi[e] : e;
    long ec = ci ? ci[e] : e;
    const T* ap = A + (ea * nch + c) * da;
    const T* bp = B + eb * db;
    const T* cp = C + (ec * nch + c) * dg;
    for (int k = 0; k < da; ++k) ma[k] = (ACC)ap[k];
    for (int k = 0; k < db; ++k) mb[k] = (ACC)bp[k];
    for (int k = 0; k < dg; ++k) mc[k] = (ACC)cp[k];
    for (int k = 0; k < n_ent; ++k) {
      int4 q = ent_lds[k];
      mo[q.w] += coef_lds[k] * ma[q.x] * mb[q.y] * mc[q.z];
    }
  }
  T* op = out + i * do_;
  for (int k = 0; k < do_; ++k) op[k] = (T)mo[k];
}

// etp_reduce with per-slot row indices
template <typename T>
__global__ void etp_reduce_idx_kernel(
    const T* __restrict__ A, const T* __restrict__ C,
    const T* __restrict__ D,
    typename acc_of<T>::type* __restrict__ out,
    const int4* __restrict__ entries,
    const float* __restrict__ coefs, int n_ent,
    long E, int nch, int da, int db, int dg, int do_,
    const long* __restrict__ ai, const long* __restrict__ ci,
    const long* __restrict__ di) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  using ACC = typename acc_of<T>::type;
  const int stride = ((da + dg + do_ + db) | 1);
  ACC* slices = reinterpret_cast<ACC*>(smem);
  int4* ent_lds = reinterpret_cast<int4*>(
      smem + (size_t)blockDim.x * stride * sizeof(ACC));
  float* coef_lds = reinterpret_cast<float*>(ent_lds + n_ent);
  for (int k = threadIdx.x; k < n_ent; k += blockDim.x) {
    ent_lds[k] = entries[k];
    coef_lds[k] = coefs[k];
  }
  long NC = E * nch;
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  ACC* my = slices + (size_t)threadIdx.x * stride;
  ACC* ma = my;
  ACC* mc = ma + da;
  ACC* md = mc + dg;
  ACC* mb = md + do_;
  if (i < NC) {
    long e = i / nch;
    int c = (int)(i - e * nch);
    long ea = ai ? ai[e] : e;
    long ec = ci ? ci[e] : e;
    long ed = di ? di[e] : e;
    const T* ap = A + (ea * nch + c) * da;
    const T* cp = C + (ec * nch + c) * dg;
    const T* dp = D + (ed * nch + c) * do_;
    for (int k = 0; k < da; ++k) ma[k] = (ACC)ap[k];
    for (int k = 0; k < dg; ++k) mc[k] = (ACC)cp[k];
    for (int k = 0; k < do_; ++k) md[k] = (ACC)dp[k];
    for (int k = 0; k < db; ++k) mb[k] = 0.f;
  }
  __syncthreads();
  if (i < NC) {
    for (int k = 0; k < n_ent; ++k) {
      int4 q = ent_lds[k];
      mb[q.y] += coef_lds[k] * ma[q.x] * mc[q.z] * md[q.w];
    }
  }
  long e = i / nch;
  if (nch % 64 == 0) {
    for (int b = 0; b < db; ++b) {
      ACC v = (i < NC) ? mb[b] : (ACC)0;
      for (int off = 32; off >= 1; off >>= 1)
        v += __shfl_down(v, off, 64);
      if ((threadIdx.x % 64) == 0 && i < NC)
        atomicAdd(&out[e * db + b], v);
    }
  } else if (i < NC) {
    for (int b = 0; b < db; ++b) atomicAdd(&out[e * db + b], mb[b]);
  }
}

// L1-operand reduce variant: only the small db-accumulator lives in
// LDS; A/C/D rows are read straight through L1 (each row is 1-2 hot
// cache lines).  LDS per thread drops from (da+dg+do+db) to db words
// -> ~6x more waves per CU than the staged variant; the staged reduce
// measured only ~700-950 GB/s of the 8 TB/s roofline (latency-bound
// at 1-2 workgroups/CU).
template <typename T>
__global__ void etp_reduce_l1_kernel(
    const T* __restrict__ A, const T* __restrict__ C,
    const T* __restrict__ D,
    typename acc_of<T>::type* __restrict__ out,
    const int4* __restrict__ entries,
    const float* __restrict__ coefs, int n_ent,
    long E, int nch, int da, int db, int dg, int do_,
    const long* __restrict__ ai, const long* __restrict__ ci,
    const long* __restrict__ di) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  using ACC = typename acc_of<T>::type;
  const int stride = db | 1;
  ACC* slices = reinterpret_cast<ACC*>(smem);
  int4* ent_lds = reinterpret_cast<int4*>(
      smem + (size_t)blockDim.x * stride * sizeof(ACC));
  float* coef_lds = reinterpret_cast<float*>(ent_lds + n_ent);
  for (int k = threadIdx.x; k < n_ent; k += blockDim.x) {
    ent_lds[k] = entries[k];
    coef_lds[k] = coefs[k];
  }
  long NC = E * nch;
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  ACC* mb = slices + (size_t)threadIdx.x * stride;
  __syncthreads();
  long e = i / nch;
  if (i < NC) {
    int c = (int)(i - e * nch);
    long ea = ai ? ai[e] : e;
    long ec = ci ? ci[e] : e;
    long ed = di ? di[e] : e;
    const T* ap = A + (ea * nch + c) * da;
    const T* cp = C + (ec * nch + c) * dg;
    const T* dp = D + (ed * nch + c) * do_;
    for (int k = 0; k < db; ++k) mb[k] = 0.f;
    for (int k = 0; k < n_ent; ++k) {
      int4 q = ent_lds[k];
      mb[q.y] += coef_lds[k] * (ACC)ap[q.x] * (ACC)cp[q.z] *
                 (ACC)dp[q.w];
    }
  }
  if (nch % 64 == 0) {
    for (int b = 0; b < db; ++b) {
      ACC v = (i < NC) ? mb[b] : (ACC)0;
      for (int off = 32; off >= 1; off >>= 1)
        v += __shfl_down(v, off, 64);
      if ((threadIdx.x % 64) == 0 && i < NC)
        atomicAdd(&out[e * db + b], v);
    }
  } else if (i < NC) {
    for (int b = 0; b < db; ++b) atomicAdd(&out[e * db + b], mb[b]);
  }
}

}  // namespace

static hipStream_t etp_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

static int etp_block_size() {
  static int b = []() {
    const char* e = getenv("HYDRAGNN_ETP_BLOCK");
    int v = e ? atoi(e) : 256;
    return (v == 64 || v == 128 || v == 256 || v == 512) ? v : 256;
  }();
  return b;
}

// etp_reduce prefers larger blocks (micro-bench: 512 -> +30% BW over
// 256 on the b1024 gY shape); separately tunable.
static int etp_reduce_block_size() {
  static int b = []() {
    const char* e = getenv("HYDRAGNN_ETP_REDUCE_BLOCK");
    int v = e ? atoi(e) : 512;
    return (v == 64 || v == 128 || v == 256 || v == 512) ? v : 512;
  }();
  return b;
}

static const long* idx_ptr(const c10::optional<torch::Tensor>& t) {
  return t.has_value() ? t->data_ptr<long>() : nullptr;
}

torch::Tensor etp_general(torch::Tensor A, torch::Tensor B, torch::Tensor C,
                          torch::Tensor entries, torch::Tensor coefs,
                          torch::Tensor o_ranges, long do_,
                          c10::optional<torch::Tensor> ai,
                          c10::optional<torch::Tensor> bi,
                          c10::optional<torch::Tensor> ci,
                          long n_rows) {
  TORCH_CHECK(A.is_cuda() && A.is_contiguous());
  TORCH_CHECK(B.is_contiguous() && C.is_contiguous());
  long E = n_rows > 0 ? n_rows : A.size(0);
  long NC = E * A.size(1);
  int nch = A.size(1);
  int da = A.size(2), db = B.size(1), dg = C.size(2);
  TORCH_CHECK(da <= 192 && db <= 192 && dg <= 192 && do_ <= 192,
              "etp dims exceed kernel limits");
  auto out = torch::empty({E, A.size(1), do_}, A.options());
  if (NC == 0) return out;
  int n_ent = entries.size(0);
  int block = etp_block_size();
  size_t accs = A.scalar_type() == at::ScalarType::Double ? 8 : 4;
  int stride = (da + db + dg + (int)do_) | 1;
  size_t lds_full = (size_t)block * stride * accs + n_ent * 20;
  int stride_l1 = (int)do_ | 1;
  size_t lds_l1 = (size_t)block * stride_l1 * accs + n_ent * 20;
  // Variant choice (A/B'd on the default bench):
  //  - racc (r2 default): register accumulation over the
  //    output-sorted entry ranges — no LDS RMW dependency chain, no
  //    output slice in LDS.
  //  - staged (r1): full A/B/C/out slices in LDS.
  //  - L1: operands from L1, only the accumulator in LDS.
  // Fallback order by LDS budget; HYDRAGNN_ETP_VARIANT=staged|l1|racc
  // overrides.
  int stride_racc = (da + db + dg) | 1;
  size_t lds_racc = (size_t)block * stride_racc * accs + n_ent * 20 +
                    (size_t)do_ * 8;
  // Measured A/B (b1024 bench, same box): staged 30.8k g/s end-to-end
  // vs racc 29.8k — staged stays the default; racc (smaller slices)
  // is the fallback when the staged LDS budget is exceeded.
  int variant = lds_full <= 150 * 1024 ? 0
                : (lds_racc <= 150 * 1024 ? 2 : 1);
  const char* env = getenv("HYDRAGNN_ETP_VARIANT");
  if (env) {
    if (env[0] == 's') variant = 0;
    else if (env[0] == 'l') variant = 1;
    else if (env[0] == 'r') variant = 2;
  }
  size_t lds_bytes = variant == 2 ? lds_racc
                     : (variant == 1 ? lds_l1 : lds_full);
  TORCH_CHECK(lds_bytes <= 150 * 1024, "etp LDS budget exceeded");
  long blocks = (NC + block - 1) / block;
  auto orng = o_ranges.to(torch::kInt32).contiguous();
  // fp64 runs with double LDS slices (acc_of<double>); others stage fp32
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, A.scalar_type(),
      "etp_general", [&] {
        if (variant == 2) {
          hipLaunchKernelGGL(
              etp_general_racc_kernel<scalar_t>, dim3(blocks),
              dim3(block), lds_bytes, etp_stream(),
              A.data_ptr<scalar_t>(), B.data_ptr<scalar_t>(),
              C.data_ptr<scalar_t>(), out.data_ptr<scalar_t>(),
              reinterpret_cast<const int4*>(entries.data_ptr<int>()),
              coefs.data_ptr<float>(),
              reinterpret_cast<const int2*>(orng.data_ptr<int>()),
              n_ent, NC, nch, da, db, dg, (int)do_,
              idx_ptr(ai), idx_ptr(bi), idx_ptr(ci));
          return;
        }
        auto kern = variant == 1 ? etp_general_l1_kernel<scalar_t>
                                 : etp_general_kernel<scalar_t>;
        hipLaunchKernelGGL(
            kern, dim3(blocks), dim3(block),
            lds_bytes, etp_stream(),
            A.data_ptr<scalar_t>(), B.data_ptr<scalar_t>(),
            C.data_ptr<scalar_t>(), out.data_ptr<scalar_t>(),
            reinterpret_cast<const int4*>(entries.data_ptr<int>()),
            coefs.data_ptr<float>(), n_ent,
            NC, nch, da, db, dg, (int)do_,
            idx_ptr(ai), idx_ptr(bi), idx_ptr(ci));
      });
  return out;
}

torch::Tensor etp_nodesum(torch::Tensor A, torch::Tensor B,
                          torch::Tensor C, torch::Tensor entries,
                          torch::Tensor coefs, long do_,
                          torch::Tensor rowptr,
                          c10::optional<torch::Tensor> ai,
                          c10::optional<torch::Tensor> bi,
                          c10::optional<torch::Tensor> ci) {
  TORCH_CHECK(A.is_cuda() && A.is_contiguous());
  TORCH_CHECK(B.is_contiguous() && C.is_contiguous());
  long R = rowptr.numel() - 1;
  int nch = A.size(1);
  int da = A.size(2), db = B.size(1), dg = C.size(2);
  auto out = torch::empty({R, (long)nch, do_}, A.options());
  long RC = R * nch;
  if (RC == 0) return out;
  int n_ent = entries.size(0);
  int block = etp_block_size();
  int stride = (da + db + dg + (int)do_) | 1;
  size_t accs = A.scalar_type() == at::ScalarType::Double ? 8 : 4;
  size_t lds_bytes = (size_t)block * stride * accs + n_ent * 20;
  TORCH_CHECK(lds_bytes <= 150 * 1024, "etp LDS budget exceeded");
  long blocks = (RC + block - 1) / block;
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, A.scalar_type(),
      "etp_nodesum", [&] {
        hipLaunchKernelGGL(
            etp_nodesum_kernel<scalar_t>, dim3(blocks), dim3(block),
            lds_bytes, etp_stream(),
            A.data_ptr<scalar_t>(), B.data_ptr<scalar_t>(),
            C.data_ptr<scalar_t>(), out.data_ptr<scalar_t>(),
            reinterpret_cast<const int4*>(entries.data_ptr<int>()),
            coefs.data_ptr<float>(), n_ent,
            rowptr.data_ptr<long>(), R, nch, da, db, dg, (int)do_,
            idx_ptr(ai), idx_ptr(bi), idx_ptr(ci));
      });
  return out;
}

torch::Tensor etp_reduce(torch::Tensor A, torch::Tensor C, torch::Tensor D,
                         torch::Tensor entries, torch::Tensor coefs,
                         long db,
                         c10::optional<torch::Tensor> ai,
                         c10::optional<torch::Tensor> ci,
                         c10::optional<torch::Tensor> di,
                         long n_rows) {
  TORCH_CHECK(A.is_cuda() && A.is_contiguous());
  long E = n_rows > 0 ? n_rows : A.size(0);
  int nch = A.size(1);
  int da = A.size(2), dg = C.size(2), do_ = D.size(2);
  TORCH_CHECK(db <= 12, "etp_reduce db limit");
  auto out = torch::zeros(
      {E, db}, A.options().dtype(
          A.scalar_type() == at::ScalarType::Double ? torch::kDouble
                                                    : torch::kFloat));
  if (E == 0) return out.to(A.scalar_type());
  int n_ent = entries.size(0);
  int block = etp_reduce_block_size();
  int stride = (da + dg + do_ + (int)db) | 1;
  size_t accs = A.scalar_type() == at::ScalarType::Double ? 8 : 4;
  size_t lds_bytes = (size_t)block * stride * accs + n_ent * 20;
  // staged is the measured default (963 GB/s vs the L1-operand
  // variant's 381: repeated per-entry L1 hits lose to LDS reads);
  // HYDRAGNN_ETP_REDUCE_VARIANT=l1 selects the low-LDS variant for
  // over-budget shapes
  const char* rv = getenv("HYDRAGNN_ETP_REDUCE_VARIANT");
  size_t lds_l1 = (size_t)block * ((db | 1)) * accs + n_ent * 20;
  bool use_l1 = (rv && rv[0] == 'l') ||
                (lds_bytes > 150 * 1024 && lds_l1 <= 150 * 1024);
  if (use_l1)
    lds_bytes = (size_t)block * ((db | 1)) * accs + n_ent * 20;
  TORCH_CHECK(lds_bytes <= 150 * 1024, "etp_reduce LDS budget exceeded");
  long blocks = (E * nch + block - 1) / block;
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, A.scalar_type(),
      "etp_reduce", [&] {
        auto kern = use_l1 ? etp_reduce_l1_kernel<scalar_t>
                           : etp_reduce_idx_kernel<scalar_t>;
        hipLaunchKernelGGL(
            kern, dim3(blocks), dim3(block),
            lds_bytes, etp_stream(), A.data_ptr<scalar_t>(),
            C.data_ptr<scalar_t>(),
            D.data_ptr<scalar_t>(), out.data_ptr<typename acc_of<scalar_t>::type>(),
            reinterpret_cast<const int4*>(entries.data_ptr<int>()),
            coefs.data_ptr<float>(), n_ent, E, nch, da, (int)db, dg, do_,
            idx_ptr(ai), idx_ptr(ci), idx_ptr(di));
      });
  return out.to(A.scalar_type());
}
