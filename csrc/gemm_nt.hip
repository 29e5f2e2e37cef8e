// NT-layout MFMA GEMM for the BERT linear layers:
// C[M, N] = A[M, K] @ B[N, K]^T (+ bias[N]) with both operands K-major
// row-major — exactly torch's F.linear(x, w): A = activations, B = the
// [out, in] weight. Both MFMA fragments read 8 contiguous bf16 along K,
// so no transpose anywhere (the layout hipBLASLt calls NT and where the
// in-tree wgrad TT kernels lose to Tensile).
//
// Structure (guide §5 "step-3" ladder stage, measured 874-912 TF/s at
// 4096^3 on this chip): 128x128 output tile per 256-thread workgroup
// (4 waves, 2x2 of 64x64), BK = 64, double-buffered LDS staged with
// global_load_lds width 16 (lane-linear image, XOR-swizzled SOURCE
// address so the MFMA fragment ds_read_b128s hit <=2-way banks),
// bijective XCD-aware workgroup remap (8 XCDs, private L2s).
//
// Full tiles only: the host wrapper dispatches here when M%128==0,
// N%128==0, K%64==0 (every BERT shape: K/N in {768, 2304, 3072},
// M = batch*seq); anything else falls back to the library GEMM.
#include "common.h"

#define BM 128
#define BN 128
#define BK 64

using bfrag = mfma_bf16x8;
using cfrag = mfma_f32x4;

__device__ __forceinline__ cfrag mfma16(bfrag a, bfrag b, cfrag c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

// Source-address chunk swizzle: the LDS image is lane-linear (glds), so
// the bank-spread XOR moves to the per-lane GLOBAL address. A row is
// 64 bf16 = 128 B = one cache line, and the XOR permutes 16-B chunks
// WITHIN the row, so coalescing is preserved exactly.
__device__ __forceinline__ int swz_chunk(int row, int chunk) {
  return chunk ^ (row & 7);
}

// Stage one [128][BK] bf16 tile (rows r0.., K-major, row stride `ld`
// elements) into lds (lane-linear [128][BK]): 4 glds-b128 per wave.
__device__ __forceinline__ void stage_tile(const bf16* __restrict__ g,
                                           long ld, long r0, long k0,
                                           bf16* lds) {
  const int t = threadIdx.x;                 // 0..255
  // one glds instruction stages 64 lanes x 16 B = 1 KiB = 8 rows
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int lds_off = (t / WAVE) * 4096 + i * 1024 + (t % WAVE) * 16;
    const int row = lds_off >> 7;            // 128 B per row
    const int chunk = (lds_off & 127) >> 4;  // 16-B chunk in row
    const long src = (r0 + row) * ld + k0 +
                     (long)(swz_chunk(row, chunk) << 3);
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)(g + src),
        (__attribute__((address_space(3))) void*)(lds + (lds_off >> 1)),
        16, 0, 0);
  }
}

// MFMA operand fragment from the lane-linear swizzled image:
// lane l -> tile[r0 + (l&15)][k0 + (l>>4)*8 .. +8]   (b128, <=2-way)
__device__ __forceinline__ bfrag frag(const bf16* tile, int r0, int k0) {
  const int l = threadIdx.x & (WAVE - 1);
  const int row = r0 + (l & 15);
  const int chunk = (k0 >> 3) + (l >> 4);
  return *reinterpret_cast<const bfrag*>(
      tile + ((long)row << 6) + (swz_chunk(row, chunk) << 3));
}

template <typename OT, bool HAS_BIAS>
__global__ __launch_bounds__(256, 2) void gemm_nt_kernel(
    const bf16* __restrict__ A,   // [M, K]
    const bf16* __restrict__ B,   // [N, K]
    const float* __restrict__ bias,  // [N] fp32 or null
    OT* __restrict__ C,           // [M, N]
    long M, long N, long K, int tiles_n) {
  __shared__ bf16 smem[2 * 2 * BM * BK];     // A/B double-buffered, 64 KiB
  bf16* a_s = smem;                          // [2][128][64]
  bf16* b_s = smem + 2 * BM * BK;

  // bijective XCD-aware remap (8 XCDs; guide T1)
  const int nwg = gridDim.x;
  int wg = blockIdx.x;
  {
    const int q = nwg >> 3, r = nwg & 7;
    const int xcd = wg & 7, idx = wg >> 3;
    wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const long m0 = (long)(wg / tiles_n) * BM;
  const long n0 = (long)(wg % tiles_n) * BN;

  const int wid = threadIdx.x / WAVE;        // 2x2 wave grid
  const int wr = (wid >> 1) * 64, wc = (wid & 1) * 64;
  const int lane = threadIdx.x & (WAVE - 1);

  cfrag acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = cfrag{0.f, 0.f, 0.f, 0.f};

  stage_tile(A, K, m0, 0, a_s);
  stage_tile(B, K, n0, 0, b_s);
  __syncthreads();                           // drains glds (vmcnt 0)

  const int nkt = (int)(K / BK);
  int buf = 0;
  for (int kt = 0; kt < nkt; ++kt) {
    if (kt + 1 < nkt) {                      // prefetch next K-tile
      const int nxt = buf ^ 1;
      stage_tile(A, K, m0, (long)(kt + 1) * BK, a_s + nxt * BM * BK);
      stage_tile(B, K, n0, (long)(kt + 1) * BK, b_s + nxt * BM * BK);
    }
    const bf16* at = a_s + buf * BM * BK;
    const bf16* bt = b_s + buf * BM * BK;
#pragma unroll
    for (int kk = 0; kk < BK / 32; ++kk) {
      bfrag af[4], bfr[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) af[i] = frag(at, wr + i * 16, kk * 32);
#pragma unroll
      for (int j = 0; j < 4; ++j) bfr[j] = frag(bt, wc + j * 16, kk * 32);
      // favor this wave while its MFMA cluster runs (guide T5)
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] = mfma16(af[i], bfr[j], acc[i][j]);
      __builtin_amdgcn_s_setprio(0);
    }
    __syncthreads();                         // joins + drains prefetch
    buf ^= 1;
  }

  // epilogue: C/D map col = lane&15, row = (lane>>4)*4 + r
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const long row0 = m0 + wr + i * 16 + ((lane >> 4) << 2);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const long col = n0 + wc + j * 16 + (lane & 15);
      const float bv = HAS_BIAS ? bias[col] : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r)
        from_f32(acc[i][j][r] + bv, &C[(row0 + r) * N + col]);
    }
  }
}

template <typename OT>
static void launch_gemm_nt(const at::Tensor& a, const at::Tensor& b,
                           const c10::optional<at::Tensor>& bias,
                           at::Tensor& out, long M, long N, long K) {
  dim3 grid((unsigned)((M / BM) * (N / BN)));
  const float* bp =
      bias ? bias->data_ptr<float>() : nullptr;
  if (bp)
    hipLaunchKernelGGL((gemm_nt_kernel<OT, true>), grid, dim3(256), 0,
                       cur_stream(a), (const bf16*)a.data_ptr(),
                       (const bf16*)b.data_ptr(), bp, (OT*)out.data_ptr(),
                       M, N, K, (int)(N / BN));
  else
    hipLaunchKernelGGL((gemm_nt_kernel<OT, false>), grid, dim3(256), 0,
                       cur_stream(a), (const bf16*)a.data_ptr(),
                       (const bf16*)b.data_ptr(), nullptr,
                       (OT*)out.data_ptr(), M, N, K, (int)(N / BN));
}

// C = A @ B^T (+bias). A [M,K] bf16, B [N,K] bf16, bias fp32 optional;
// out bf16 (fp32_out=false) or fp32.
at::Tensor gemm_nt(const at::Tensor& a, const at::Tensor& b,
                   c10::optional<at::Tensor> bias, bool fp32_out) {
  CHECK_CUDA_CONTIG(a);
  CHECK_CUDA_CONTIG(b);
  TORCH_CHECK(a.scalar_type() == at::kBFloat16 &&
                  b.scalar_type() == at::kBFloat16,
              "gemm_nt: bf16 operands");
  const long M = a.size(0), K = a.size(1), N = b.size(0);
  TORCH_CHECK(b.size(1) == K, "gemm_nt: K mismatch");
  TORCH_CHECK(M % BM == 0 && N % BN == 0 && K % BK == 0,
              "gemm_nt: needs M%128==0, N%128==0, K%64==0 (got ", M, "x", N,
              "x", K, ")");
  if (bias) {
    TORCH_CHECK(bias->is_cuda() && bias->is_contiguous() &&
                    bias->scalar_type() == at::kFloat &&
                    bias->numel() == N,
                "gemm_nt: bias must be fp32 [N]");
  }
  auto out = at::empty({M, N}, a.options().dtype(
                                   fp32_out ? at::kFloat : at::kBFloat16));
  if (fp32_out)
    launch_gemm_nt<float>(a, b, bias, out, M, N, K);
  else
    launch_gemm_nt<bf16>(a, b, bias, out, M, N, K);
  HIP_CHECK_LAST();
  return out;
}
