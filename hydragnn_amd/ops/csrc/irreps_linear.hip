// Per-l channel-mixing linear on irreps towers (gfx950).
//
// out[n, co, m] = sum_ci x[n, ci, m] * W[lmap[m], ci, co]  (+ bias on
// m = 0).  This is the o3.Linear equivalent on the dense
// uniform-multiplicity layout [N, C, D], D = (lmax+1)^2 — the most
// common GEMM family in MACE after the tensor products.
//
// The torch path (bmm over D slices) costs 3x HBM traffic: permute
// copy -> D skinny hipBLASLt GEMMs (measured 1.4-3% MFMA issue
// density) -> permute-back copy.  Here one workgroup owns a BM-row
// n-tile: it reads x[n0:n0+16, :, :] once at full coalescing,
// transposes it into LDS as [m][n][ci], keeps the whole weight stack
// [L, Cin, Cout] in LDS, runs all D 16xCout GEMMs on
// v_mfma_f32_16x16x32_bf16 with every m accumulated in registers, and
// writes out[n, :, :] once.  Single-pass traffic, no library calls.
//
// TRANS_W reads W as [L, Cout, Cin] (the gX backward with swapped
// roles), so the family is closed under differentiation and force
// training stays on this kernel.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

namespace {

using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x4 = __attribute__((ext_vector_type(4))) float;

constexpr int BM = 32;       // n rows per workgroup
constexpr int MAXD = 16;     // (lmax+1)^2, lmax <= 3
constexpr int PAD = 8;       // bf16 elems of row padding (16 B)

template <bool TRANS_W>
__global__ __launch_bounds__(256) void irreps_linear_kernel(
    const __hip_bfloat16* __restrict__ X,   // [N, Cin, D]
    const __hip_bfloat16* __restrict__ W,   // [L, Cin, Cout] ([L,Cout,Cin] if TRANS_W)
    const float* __restrict__ bias,         // [Cout] or nullptr (m=0 only)
    const __hip_bfloat16* __restrict__ add, // [N, Cout, D] residual or nullptr
    __hip_bfloat16* __restrict__ out,       // [N, Cout, D]
    const long* __restrict__ lmap,          // [D] -> l index
    long N, int Cin, int Cout, int D, int L) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int CP = Cin + PAD;    // lA row stride
  const int WP = Cout + PAD;   // lW row stride
  __hip_bfloat16* lA = reinterpret_cast<__hip_bfloat16*>(smem);  // [D][BM][CP]
  __hip_bfloat16* lW = lA + D * BM * CP;         // [L][Cin][WP]
  __shared__ int lmap_s[MAXD];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const long n0 = (long)blockIdx.x * BM;

  if (tid < D) lmap_s[tid] = (int)lmap[tid];

  // stage x[n0:n0+16, :, :]: contiguous int4 global loads, scalar
  // LDS writes doing the [n][ci][m] -> [m][n][ci] transpose
  {
    const int row_elems = Cin * D;               // per n row, % 8 == 0
    const int chunks = BM * row_elems / 8;
    for (int idx = tid; idx < chunks; idx += 256) {
      int r = idx / (row_elems / 8);
      int f = (idx - r * (row_elems / 8)) * 8;   // flat (ci, m) offset
      long gn = n0 + r;
      if (gn < N) {
        bf16x8 v = *reinterpret_cast<const bf16x8*>(
            X + gn * row_elems + f);
        for (int t = 0; t < 8; ++t) {
          int ci = (f + t) / D, m = (f + t) - ((f + t) / D) * D;
          lA[(m * BM + r) * CP + ci] = __hip_bfloat16(v[t]);
        }
      } else {
        for (int t = 0; t < 8; ++t) {
          int ci = (f + t) / D, m = (f + t) - ((f + t) / D) * D;
          lA[(m * BM + r) * CP + ci] = __hip_bfloat16(0.f);
        }
      }
    }
  }
  // stage W -> lW[l][ci][co] (padded rows)
  if (!TRANS_W) {
    const int chunks = L * Cin * Cout / 8;
    for (int idx = tid; idx < chunks; idx += 256) {
      int row = idx / (Cout / 8);                // l*Cin + ci
      int co = (idx - row * (Cout / 8)) * 8;
      *reinterpret_cast<bf16x8*>(&lW[row * WP + co]) =
          *reinterpret_cast<const bf16x8*>(W + (long)row * Cout + co);
    }
  } else {
    // W given [L, Cout, Cin]: strided gather (panel is tiny)
    for (int idx = tid; idx < L * Cin * Cout; idx += 256) {
      int l = idx / (Cin * Cout);
      int rem = idx - l * Cin * Cout;
      int ci = rem / Cout, co = rem - (rem / Cout) * Cout;
      lW[(l * Cin + ci) * WP + co] =
          W[((long)l * Cout + co) * Cin + ci];
    }
  }
  __syncthreads();

  // waves round-robin the (row-tile, co-tile) grid of 16x16 sub-tiles
  const int nct = Cout / 16;
  for (int t = wave; t < (BM / 16) * nct; t += 4) {
    const int rt = t / nct;
    const int ct = t - rt * nct;
    f32x4 acc[MAXD];
    for (int m = 0; m < D; ++m) acc[m] = {0.f, 0.f, 0.f, 0.f};
    for (int k0 = 0; k0 < Cin; k0 += 32) {
      for (int m = 0; m < D; ++m) {
        // A fragment: lane holds lA[m][row][k0 + (lane>>4)*8 + j]
        const __hip_bfloat16* ap =
            &lA[(m * BM + rt * 16 + (lane & 15)) * CP + k0 +
                (lane >> 4) * 8];
        bf16x8 afrag = *reinterpret_cast<const bf16x8*>(ap);
        // B fragment: lane holds lW[l][k0+(lane>>4)*8+q][ct*16+(lane&15)]
        const __hip_bfloat16* bp =
            &lW[(lmap_s[m] * Cin + k0 + (lane >> 4) * 8) * WP +
                ct * 16 + (lane & 15)];
        bf16x8 bfrag;
        for (int q = 0; q < 8; ++q) bfrag[q] = (__bf16)bp[q * WP];
        acc[m] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afrag, bfrag, acc[m], 0, 0, 0);
      }
    }
    // epilogue: C layout col = lane&15, row = (lane>>4)*4 + reg;
    // optional fused residual add (saves a full elementwise pass)
    int col = ct * 16 + (lane & 15);
    float b = bias != nullptr ? bias[col] : 0.f;
    for (int reg = 0; reg < 4; ++reg) {
      long row = n0 + rt * 16 + (lane >> 4) * 4 + reg;
      if (row < N) {
        long off = (row * Cout + col) * D;
        __hip_bfloat16* op = out + off;
        if (add != nullptr) {
          const __hip_bfloat16* rp = add + off;
          for (int m = 0; m < D; ++m)
            op[m] = __float2bfloat16(acc[m][reg] + (m == 0 ? b : 0.f) +
                                     __bfloat162float(rp[m]));
        } else {
          for (int m = 0; m < D; ++m)
            op[m] = __float2bfloat16(acc[m][reg] + (m == 0 ? b : 0.f));
        }
      }
    }
  }
}

// Weight gradient: gW[l, ci, co] = sum_{n, m in l} x[n,ci,m]*g[n,co,m].
// The bmm route needs permuted COPIES of both x and g (50 MB/call at
// bench shapes); here persistent blocks stride the n-tiles, stage x
// and g through LDS with the same transpose as the forward, run the
// contraction over n on MFMA (k = n-chunk of 32), and write per-block
// fp32 partials (summed by the caller) — deterministic, no atomics,
// single-pass traffic.
__global__ __launch_bounds__(256) void irreps_linear_gw_kernel(
    const __hip_bfloat16* __restrict__ X,   // [N, Cin, D]
    const __hip_bfloat16* __restrict__ G,   // [N, Cout, D]
    float* __restrict__ partials,           // [nblocks, L, Cin, Cout]
    const long* __restrict__ lmap,          // [D]
    long N, int Cin, int Cout, int D, int L) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int CP = Cin + PAD;
  const int GP = Cout + PAD;
  __hip_bfloat16* lA = reinterpret_cast<__hip_bfloat16*>(smem);  // [D][BM][CP]
  __hip_bfloat16* lG = lA + D * BM * CP;                         // [D][BM][GP]
  __shared__ int lmap_s[MAXD];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  if (tid < D) lmap_s[tid] = (int)lmap[tid];

  // each wave owns (Cin/16 * Cout/16 / 4) sub-tiles, acc per (tile, l)
  const int nct = Cout / 16;
  const int ntiles = (Cin / 16) * nct;
  const int mytiles = ntiles / 4;             // Cin,Cout >= 32 -> >= 1
  f32x4 acc[16];                              // [mytile * L], L*mytiles<=16
  for (int i = 0; i < 16; ++i) acc[i] = {0.f, 0.f, 0.f, 0.f};

  const long tiles_n = (N + BM - 1) / BM;
  for (long tb = blockIdx.x; tb < tiles_n; tb += gridDim.x) {
    const long n0 = tb * BM;
    __syncthreads();
    {  // stage x and g tiles (transpose to [m][n][c])
      const int rex = Cin * D, reg_ = Cout * D;
      const int cx = BM * rex / 8, cg = BM * reg_ / 8;
      for (int idx = tid; idx < cx + cg; idx += 256) {
        bool isx = idx < cx;
        int id2 = isx ? idx : idx - cx;
        int re = isx ? rex : reg_;
        int P = isx ? CP : GP;
        __hip_bfloat16* dst = isx ? lA : lG;
        const __hip_bfloat16* src = isx ? X : G;
        int r = id2 / (re / 8);
        int f = (id2 - r * (re / 8)) * 8;
        long gn = n0 + r;
        if (gn < N) {
          bf16x8 v = *reinterpret_cast<const bf16x8*>(
              src + gn * re + f);
          for (int t = 0; t < 8; ++t) {
            int c = (f + t) / D, m = (f + t) - ((f + t) / D) * D;
            dst[(m * BM + r) * P + c] = __hip_bfloat16(v[t]);
          }
        } else {
          for (int t = 0; t < 8; ++t) {
            int c = (f + t) / D, m = (f + t) - ((f + t) / D) * D;
            dst[(m * BM + r) * P + c] = __hip_bfloat16(0.f);
          }
        }
      }
    }
    __syncthreads();
    for (int mt = 0; mt < mytiles; ++mt) {
      int t = wave * mytiles + mt;
      int cit = t / nct, cot = t - (t / nct) * nct;
      for (int m = 0; m < D; ++m) {
        int l = lmap_s[m];
        // k = n over the whole BM=32 tile in ONE mfma: lane group
        // (lane>>4) covers k = (lane>>4)*8 + q, q = 0..7
        // A[ci][k=n] from lA[m][n][ci] (stride-CP gather)
        const __hip_bfloat16* ap =
            &lA[(m * BM + (lane >> 4) * 8) * CP + cit * 16 + (lane & 15)];
        bf16x8 afrag;
        for (int q = 0; q < 8; ++q) afrag[q] = (__bf16)ap[q * CP];
        // B[k=n][co] from lG[m][n][co]
        const __hip_bfloat16* bp =
            &lG[(m * BM + (lane >> 4) * 8) * GP + cot * 16 + (lane & 15)];
        bf16x8 bfrag;
        for (int q = 0; q < 8; ++q) bfrag[q] = (__bf16)bp[q * GP];
        acc[mt * 4 + l] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afrag, bfrag, acc[mt * 4 + l], 0, 0, 0);
      }
    }
  }
  // flush: C layout col = lane&15 (co), row = (lane>>4)*4 + reg (ci)
  for (int mt = 0; mt < mytiles; ++mt) {
    int t = wave * mytiles + mt;
    int cit = t / nct, cot = t - (t / nct) * nct;
    for (int l = 0; l < L; ++l) {
      f32x4 a = acc[mt * 4 + l];
      for (int reg_ = 0; reg_ < 4; ++reg_) {
        int ci = cit * 16 + (lane >> 4) * 4 + reg_;
        int co = cot * 16 + (lane & 15);
        partials[(((long)blockIdx.x * L + l) * Cin + ci) * Cout + co] =
            a[reg_];
      }
    }
  }
}

}  // namespace

torch::Tensor irreps_linear(torch::Tensor X, torch::Tensor W,
                            torch::Tensor lmap,
                            c10::optional<torch::Tensor> bias,
                            bool trans_w,
                            c10::optional<torch::Tensor> add) {
  TORCH_CHECK(X.is_cuda() && X.is_contiguous());
  TORCH_CHECK(W.is_cuda() && W.is_contiguous());
  TORCH_CHECK(X.scalar_type() == at::ScalarType::BFloat16 &&
              W.scalar_type() == at::ScalarType::BFloat16,
              "irreps_linear is bf16");
  long N = X.size(0);
  int Cin = X.size(1), D = X.size(2);
  int L = W.size(0);
  int Cout = trans_w ? W.size(1) : W.size(2);
  int Cw = trans_w ? W.size(2) : W.size(1);
  TORCH_CHECK(Cin == Cw, "channel mismatch");
  TORCH_CHECK(Cin % 32 == 0 && Cout % 64 == 0,
              "irreps_linear needs Cin % 32 == 0, Cout % 64 == 0");
  TORCH_CHECK(D <= 16 && lmap.numel() == D);
  size_t lds_bytes = ((size_t)D * BM * (Cin + 8) +
                      (size_t)L * Cin * (Cout + 8)) * 2;
  TORCH_CHECK(lds_bytes <= 160 * 1024, "irreps_linear LDS budget");
  auto out = torch::empty({N, (long)Cout, (long)D}, X.options());
  if (N == 0) return out;
  const float* bias_ptr = nullptr;
  torch::Tensor bias_f;
  if (bias.has_value()) {
    bias_f = bias->to(torch::kFloat).contiguous();
    bias_ptr = bias_f.data_ptr<float>();
  }
  const __hip_bfloat16* add_ptr = nullptr;
  torch::Tensor add_c;
  if (add.has_value()) {
    add_c = add->contiguous();
    TORCH_CHECK(add_c.sizes() == out.sizes() &&
                add_c.scalar_type() == at::ScalarType::BFloat16,
                "irreps_linear residual shape/dtype mismatch");
    add_ptr = reinterpret_cast<const __hip_bfloat16*>(add_c.data_ptr());
  }
  auto lmap_c = lmap.contiguous();
  dim3 grid((N + BM - 1) / BM);
  auto stream = at::hip::getCurrentHIPStream().stream();
  auto launch = [&](auto kern) {
    hipLaunchKernelGGL(kern, grid, dim3(256), lds_bytes, stream,
                       reinterpret_cast<const __hip_bfloat16*>(X.data_ptr()),
                       reinterpret_cast<const __hip_bfloat16*>(W.data_ptr()),
                       bias_ptr, add_ptr,
                       reinterpret_cast<__hip_bfloat16*>(out.data_ptr()),
                       lmap_c.data_ptr<long>(), N, Cin, Cout, D, L);
  };
  if (trans_w) launch(irreps_linear_kernel<true>);
  else launch(irreps_linear_kernel<false>);
  return out;
}

torch::Tensor irreps_linear_gw(torch::Tensor X, torch::Tensor G,
                               torch::Tensor lmap, long L,
                               long nblocks) {
  TORCH_CHECK(X.is_cuda() && X.is_contiguous());
  TORCH_CHECK(G.is_cuda() && G.is_contiguous());
  TORCH_CHECK(X.scalar_type() == at::ScalarType::BFloat16 &&
              G.scalar_type() == at::ScalarType::BFloat16);
  long N = X.size(0);
  int Cin = X.size(1), D = X.size(2);
  int Cout = G.size(1);
  TORCH_CHECK(G.size(0) == N && G.size(2) == D);
  int tiles = (Cin / 16) * (Cout / 16);
  TORCH_CHECK(Cin % 16 == 0 && Cout % 16 == 0 && tiles % 4 == 0 &&
              tiles <= 16 && L <= 4 && D <= 16,
              "irreps_linear_gw shape envelope");
  size_t lds_bytes = ((size_t)D * BM * (Cin + 8) +
                      (size_t)D * BM * (Cout + 8)) * 2;
  TORCH_CHECK(lds_bytes <= 160 * 1024, "irreps_linear_gw LDS budget");
  auto partials = torch::empty(
      {nblocks, L, Cin, Cout},
      X.options().dtype(torch::kFloat));
  if (N == 0) return partials.zero_();
  auto lmap_c = lmap.contiguous();
  auto stream = at::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(irreps_linear_gw_kernel, dim3(nblocks), dim3(256),
                     lds_bytes, stream,
                     reinterpret_cast<const __hip_bfloat16*>(X.data_ptr()),
                     reinterpret_cast<const __hip_bfloat16*>(G.data_ptr()),
                     partials.data_ptr<float>(),
                     lmap_c.data_ptr<long>(), N, Cin, Cout, D, (int)L);
  return partials;
}

