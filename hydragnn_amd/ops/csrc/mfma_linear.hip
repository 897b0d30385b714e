// bf16 MFMA GEMM for per-edge MLP layers (gfx950).
//
// Shapes of interest: C[M, N] = A[M, K] @ B[K, N] with M = num edges
// (1e5-1e6) and N, K in [32, 512] — the MACE radial MLP and readout
// layers.  hipBLASLt runs these skinny shapes ~15x off the memory
// roofline; this kernel streams A once at full coalescing, keeps the
// whole B panel in LDS, and accumulates on the matrix cores
// (v_mfma_f32_16x16x32_bf16, fp32 accumulate).
//
// Tile: 256-thread block = 4 waves; block tile BM=128 x BN=64
// (2x2 wave grid, each wave one 64x32 sub-tile of 4x2 16x16 MFMA
// fragments); K-loop in 32-deep steps with A staged through LDS
// (+8-byte row pad against bank conflicts), B staged once per K-step.
// transpose_b=true reads B as W[N, K] (the dX = g @ W^T backward) at
// identical cost.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

namespace {

using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x4 = __attribute__((ext_vector_type(4))) float;

constexpr int BM = 128;
constexpr int BN = 64;
constexpr int BK = 32;
// LDS rows padded: 32 bf16 = 64 B per row + 8 B pad -> stride 36 elems
constexpr int APAD = 36;
constexpr int BPAD = 72;  // B rows are 64 bf16 = 128 B + 16 B pad

template <bool TRANS_B>
__global__ __launch_bounds__(256) void mfma_linear_kernel(
    const __hip_bfloat16* __restrict__ A,   // [M, K]
    const __hip_bfloat16* __restrict__ B,   // [K, N] or [N, K]
    const float* __restrict__ bias,         // [N] or nullptr
    __hip_bfloat16* __restrict__ C,         // [M, N]
    long M, int N, int K) {
  __shared__ __hip_bfloat16 lA[BM * APAD];
  __shared__ __hip_bfloat16 lB[BK * BPAD];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;          // 0..3 -> (wr, wc) = (wave>>1, wave&1)
  const int wr = wave >> 1;
  const int wc = wave & 1;
  const long m0 = (long)blockIdx.x * BM;
  const int n0 = blockIdx.y * BN;

  // accumulators: 4 (m) x 2 (n) fragments of 16x16
  f32x4 acc[4][2];
  for (int i = 0; i < 4; ++i)
    for (int j = 0; j < 2; ++j)
      acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  for (int k0 = 0; k0 < K; k0 += BK) {
    // stage A tile [BM, BK]: 256 threads x 16 B = 4096 B per pass;
    // tile is BM*BK*2 = 8192 B -> 2 passes, each thread 8 bf16
    for (int p = 0; p < 2; ++p) {
      int idx = p * 256 + tid;          // covers BM*BK/8 = 512 chunks
      int row = idx >> 2;               // BK/8 = 4 chunks per row
      int col8 = (idx & 3) * 8;
      long gm = m0 + row;
      const __hip_bfloat16* src = A + gm * K + k0 + col8;
      __hip_bfloat16* dst = &lA[row * APAD + col8];
      if (gm < M) {
        *reinterpret_cast<int4*>(dst) =
            *reinterpret_cast<const int4*>(src);
      } else {
        int4 z = {0, 0, 0, 0};
        *reinterpret_cast<int4*>(dst) = z;
      }
    }
    // stage B tile [BK, BN] (bf16): BK*BN*2 = 4096 B -> 1 pass,
    // each thread 8 bf16
    {
      int idx = tid;                    // 256 chunks of 8
      int row = idx >> 3;               // BN/8 = 8 chunks per row
      int col8 = (idx & 7) * 8;
      __hip_bfloat16* dst = &lB[row * BPAD + col8];
      if (!TRANS_B) {
        const __hip_bfloat16* src = B + (long)(k0 + row) * N + n0 + col8;
        *reinterpret_cast<int4*>(dst) =
            *reinterpret_cast<const int4*>(src);
      } else {
        // B is [N, K]: gather a column strip (strided reads; the
        // B panel is tiny and L2-resident)
        for (int j = 0; j < 8; ++j)
          dst[j] = B[(long)(n0 + col8 + j) * K + k0 + row];
      }
    }
    __syncthreads();

    // MFMA over the tile: wave (wr, wc) computes rows
    // [wr*64, wr*64+64) x cols [wc*32, wc*32+32)
    // A fragment (16x32): lane l holds A[l&15][(l>>4)*8 + j]
    // B fragment (32x16): lane l holds B[(l>>4)*8 + j][l&15]
    for (int i = 0; i < 4; ++i) {        // 4 m-fragments of 16 rows
      int arow = wr * 64 + i * 16 + (lane & 15);
      const __hip_bfloat16* ap = &lA[arow * APAD + (lane >> 4) * 8];
      bf16x8 afrag = *reinterpret_cast<const bf16x8*>(ap);
      for (int j = 0; j < 2; ++j) {      // 2 n-fragments of 16 cols
        int bcol = wc * 32 + j * 16 + (lane & 15);
        bf16x8 bfrag;
        const __hip_bfloat16* bp = &lB[((lane >> 4) * 8) * BPAD + bcol];
        for (int q = 0; q < 8; ++q) bfrag[q] = (__bf16)bp[q * BPAD];
        acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afrag, bfrag, acc[i][j], 0, 0, 0);
      }
    }
    __syncthreads();
  }

  // epilogue: C/D layout col = lane&15, row = (lane>>4)*4 + reg
  for (int i = 0; i < 4; ++i) {
    for (int j = 0; j < 2; ++j) {
      int col = n0 + wc * 32 + j * 16 + (lane & 15);
      float b = (bias != nullptr && col < N) ? bias[col] : 0.f;
      for (int reg = 0; reg < 4; ++reg) {
        long row = m0 + wr * 64 + i * 16 + (lane >> 4) * 4 + reg;
        if (row < M && col < N) {
          C[row * N + col] = __float2bfloat16(acc[i][j][reg] + b);
        }
      }
    }
  }
}

}  // namespace

torch::Tensor mfma_linear(torch::Tensor A, torch::Tensor B,
                          c10::optional<torch::Tensor> bias,
                          bool trans_b) {
  TORCH_CHECK(A.is_cuda() && A.is_contiguous());
  TORCH_CHECK(B.is_cuda() && B.is_contiguous());
  TORCH_CHECK(A.scalar_type() == at::ScalarType::BFloat16);
  TORCH_CHECK(B.scalar_type() == at::ScalarType::BFloat16);
  long M = A.size(0);
  int K = A.size(1);
  int N = trans_b ? B.size(0) : B.size(1);
  int Kb = trans_b ? B.size(1) : B.size(0);
  TORCH_CHECK(K == Kb, "inner dims mismatch");
  TORCH_CHECK(K % BK == 0 && N % 16 == 0,
              "mfma_linear needs K % 32 == 0 and N % 16 == 0");
  TORCH_CHECK(N % BN == 0, "mfma_linear needs N % 64 == 0");
  auto C = torch::empty({M, (long)N}, A.options());
  if (M == 0) return C;
  const float* bias_ptr = nullptr;
  torch::Tensor bias_f;
  if (bias.has_value()) {
    bias_f = bias->to(torch::kFloat).contiguous();
    bias_ptr = bias_f.data_ptr<float>();
  }
  dim3 grid((M + BM - 1) / BM, N / BN);
  auto stream = at::hip::getCurrentHIPStream().stream();
  if (trans_b) {
    hipLaunchKernelGGL((mfma_linear_kernel<true>), grid, dim3(256), 0,
                       stream,
                       reinterpret_cast<const __hip_bfloat16*>(A.data_ptr()),
                       reinterpret_cast<const __hip_bfloat16*>(B.data_ptr()),
                       bias_ptr,
                       reinterpret_cast<__hip_bfloat16*>(C.data_ptr()),
                       M, N, K);
  } else {
    hipLaunchKernelGGL((mfma_linear_kernel<false>), grid, dim3(256), 0,
                       stream,
                       reinterpret_cast<const __hip_bfloat16*>(A.data_ptr()),
                       reinterpret_cast<const __hip_bfloat16*>(B.data_ptr()),
                       bias_ptr,
                       reinterpret_cast<__hip_bfloat16*>(C.data_ptr()),
                       M, N, K);
  }
  return C;
}
