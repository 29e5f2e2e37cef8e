// Weight-gradient GEMM: dW[N, K_in] = dY^T @ X, with dY [T, N] and
// X [T, K_in] (T = tokens = contraction dim, e.g. 8192 for bs64xL128).
//
// hipBLASLt's heuristic runs these K-bound shapes at ~410 TF/s with a
// pathological split-K choice; this kernel owns the shape family
// directly: 128x128 output tiles, K split across workgroups (grid.z),
// both operands transposed-staged through LDS with bank-conflict-free
// strides (ld % 16 == 8), MFMA 16x16x32 bf16, fp32 atomic accumulation
// into a workspace that the Python wrapper casts to the weight dtype.
//
// Tile anatomy per workgroup (4 waves):
//   A = dY^T tile [128 m][64 k]   (m = output rows = dY columns)
//   B = X^T  tile [128 n][64 k]   (n = output cols = X columns)
//   each wave owns a 32x128 slab of C: m-frags {wave*2, wave*2+1},
//   all 8 n-frags -> 16 accumulators (64 VGPRs).
#include "common.h"

using bfrag = mfma_bf16x8;
using cfrag = mfma_f32x4;

#define WG_TILE 128
#define K_CHUNK 64
#define LDS_LD (K_CHUNK + 8)  // 72 elems: 16B groups land on odd banks

__device__ __forceinline__ bfrag wg_frag(const bf16* base, int i0, int k0) {
  const int l = threadIdx.x & (WAVE - 1);
  return *reinterpret_cast<const bfrag*>(base + (long)(i0 + (l & 15)) * LDS_LD +
                                         k0 + ((l >> 4) << 3));
}

// stage a [rows=128][K_CHUNK] transposed tile: global src [T, width] with
// row stride `stride`, reading rows k0..k0+K_CHUNK (clamped to kend) and
// columns c0..c0+128 -> LDS [128][LDS_LD] as tile[c][k].
__device__ __forceinline__ void stage_T(const bf16* src, long stride, int c0,
                                        int width, long k0, long kend,
                                        bf16* lds) {
  // each thread loads 8 consecutive columns of one k-row, then scatters
  // into LDS transposed (8 scalar LDS writes)
  const int per_row = WG_TILE / 8;            // 16 vector loads per k-row
  for (int i = threadIdx.x; i < K_CHUNK * per_row; i += blockDim.x) {
    const int kk = i / per_row;
    const int cc = (i % per_row) * 8;
    const long k = k0 + kk;
    s16x8 v{};
    if (k < kend && c0 + cc + 7 < width)
      v = *reinterpret_cast<const s16x8*>(src + k * stride + c0 + cc);
#pragma unroll
    for (int e = 0; e < 8; ++e)
      lds[(long)(cc + e) * LDS_LD + kk] = reinterpret_cast<const bf16*>(&v)[e];
  }
}

__global__ __launch_bounds__(256) void wgrad_kernel(
    const bf16* __restrict__ dy,  // [T, N]
    const bf16* __restrict__ x,   // [T, Kin]
    float* __restrict__ ws,       // [N, Kin] zeroed fp32 workspace
    long T, int N, int Kin, int tiles_n, int ksplit) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  bf16* a_s = reinterpret_cast<bf16*>(smem_raw);       // [128][LDS_LD] x2
  bf16* b_s = a_s + 2 * WG_TILE * LDS_LD;              // [128][LDS_LD] x2
  const int tile = blockIdx.x;
  const int tm = tile / tiles_n;
  const int tn = tile - tm * tiles_n;
  const int m0 = tm * WG_TILE;
  const int n0 = tn * WG_TILE;
  const long kslice = (T + ksplit - 1) / ksplit;
  const long kbeg = (long)blockIdx.z * kslice;
  const long kend = min(T, kbeg + kslice);
  if (kbeg >= kend) return;
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);

  cfrag acc[2][8];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[i][j] = cfrag{0.f, 0.f, 0.f, 0.f};

  // double-buffered K loop
  int buf = 0;
  stage_T(dy, N, m0, N, kbeg, kend, a_s);
  stage_T(x, Kin, n0, Kin, kbeg, kend, b_s);
  __syncthreads();
  for (long k0 = kbeg; k0 < kend; k0 += K_CHUNK) {
    const int cur = buf;
    if (k0 + K_CHUNK < kend) {
      const int nxt = 1 - buf;
      stage_T(dy, N, m0, N, k0 + K_CHUNK, kend, a_s + nxt * WG_TILE * LDS_LD);
      stage_T(x, Kin, n0, Kin, k0 + K_CHUNK, kend,
              b_s + nxt * WG_TILE * LDS_LD);
      buf = nxt;
    }
    const bf16* a = a_s + cur * WG_TILE * LDS_LD;
    const bf16* b = b_s + cur * WG_TILE * LDS_LD;
#pragma unroll
    for (int kk = 0; kk < K_CHUNK / 32; ++kk) {
#pragma unroll
      for (int i = 0; i < 2; ++i) {
        const bfrag af = wg_frag(a, (wid * 2 + i) * 16, kk * 32);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af, wg_frag(b, j * 16, kk * 32), acc[i][j], 0, 0, 0);
      }
    }
    __syncthreads();
  }

  // accumulate into the fp32 workspace (one atomic per element; ksplit
  // workgroups contend per tile)
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    const int mrow0 = m0 + (wid * 2 + i) * 16 + ((lane >> 4) << 2);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int col = n0 + j * 16 + (lane & 15);
      if (col >= Kin) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = mrow0 + r;
        if (row < N) atomicAdd(&ws[(long)row * Kin + col], acc[i][j][r]);
      }
    }
  }
}

// dW = dY^T X. Returns fp32 [N, Kin] (wrapper casts to the weight dtype).
at::Tensor wgrad(const at::Tensor& dy, const at::Tensor& x, long ksplit) {
  CHECK_CUDA_CONTIG(dy);
  CHECK_CUDA_CONTIG(x);
  TORCH_CHECK(dy.scalar_type() == at::kBFloat16 &&
                  x.scalar_type() == at::kBFloat16,
              "wgrad: bf16 operands");
  const long T = dy.size(0);
  const int N = dy.size(1);
  const int Kin = x.size(1);
  TORCH_CHECK(x.size(0) == T, "wgrad: token dims differ");
  TORCH_CHECK(N % 8 == 0 && Kin % 8 == 0, "wgrad: dims % 8");
  auto ws = at::zeros({(long)N, (long)Kin},
                      dy.options().dtype(at::kFloat));
  const int tiles_m = (N + WG_TILE - 1) / WG_TILE;
  const int tiles_n = (Kin + WG_TILE - 1) / WG_TILE;
  if (ksplit <= 0) {
    // aim for ~2 waves of 256-CU occupancy
    const long want = 512;
    ksplit = std::max<long>(1, want / std::max(1, tiles_m * tiles_n));
    ksplit = std::min<long>(ksplit, (T + K_CHUNK - 1) / K_CHUNK);
  }
  const size_t smem = (size_t)4 * WG_TILE * LDS_LD * sizeof(bf16);
  dim3 grid(tiles_m * tiles_n, 1, (unsigned)ksplit);
  hipLaunchKernelGGL(wgrad_kernel, grid, dim3(256), smem, cur_stream(dy),
                     (const bf16*)dy.data_ptr(), (const bf16*)x.data_ptr(),
                     ws.data_ptr<float>(), T, N, Kin, tiles_n, ksplit);
  HIP_CHECK_LAST();
  return ws;
}
