// Weight-gradient GEMM (EXPERIMENTAL, not on the hot path): dW[N, K_in]
// = dY^T @ X with dY [T, N], X [T, K_in] (T = tokens = contraction dim).
//
// Status: exact (unit-tested vs fp32 matmul) but ~185 TF/s vs
// hipBLASLt's ~440 TF/s on the BERT wgrad shapes, so autograd keeps the
// library GEMM. The contraction dim T is the row index of both global
// operands, so MFMA K-contiguous fragments require a transpose
// somewhere: v1 transposed during staging (scalar LDS stores dominated,
// 110 TF/s); v2 (this code) stages [K][N]-oriented tiles with fully
// vectorized copies and pays 8 scalar LDS reads per fragment instead
// (185 TF/s). Beating Tensile here needs the assembly-grade
// buffer_load + ds_write-swizzle transpose pipeline — future work;
// kept as a measured baseline for it.
//
// Tile anatomy per workgroup (4 waves): 128x128 output tile, K split
// across grid.z with fp32 atomic accumulation into a workspace.
#include "common.h"

using bfrag = mfma_bf16x8;
using cfrag = mfma_f32x4;

#define WG_TILE 128
#define K_CHUNK 64
#define LDS_N (WG_TILE + 8)  // row stride of the [K_CHUNK][128] tiles

// fragment from a [K][N]-oriented LDS tile: element (row i0+(l&15), k)
// = tile[k][i0 + (l&15)] — 8 scalar LDS reads per fragment, but staging
// stays fully vectorized (no transpose).
__device__ __forceinline__ bfrag wg_fragT(const bf16* tile, int i0, int k0) {
  const int l = threadIdx.x & (WAVE - 1);
  const int col = i0 + (l & 15);
  const int kb = k0 + ((l >> 4) << 3);
  bfrag f;
#pragma unroll
  for (int e = 0; e < 8; ++e)
    reinterpret_cast<bf16*>(&f)[e] = tile[(long)(kb + e) * LDS_N + col];
  return f;
}

// stage [K_CHUNK][128] slab of a [T, width] tensor (rows k0.., cols c0..)
// into LDS row-major — vectorized loads AND stores.
__device__ __forceinline__ void stage_nt(const bf16* src, long stride, int c0,
                                         int width, long k0, long kend,
                                         bf16* lds) {
  const int per_row = WG_TILE / 8;            // 16 vector ops per k-row
  for (int i = threadIdx.x; i < K_CHUNK * per_row; i += blockDim.x) {
    const int kk = i / per_row;
    const int cc = (i % per_row) * 8;
    const long k = k0 + kk;
    s16x8 v{};
    if (k < kend && c0 + cc + 7 < width)
      v = *reinterpret_cast<const s16x8*>(src + k * stride + c0 + cc);
    *reinterpret_cast<s16x8*>(lds + (long)kk * LDS_N + cc) = v;
  }
}

__global__ __launch_bounds__(256) void wgrad_kernel(
    const bf16* __restrict__ dy,  // [T, N]
    const bf16* __restrict__ x,   // [T, Kin]
    float* __restrict__ ws,       // [N, Kin] zeroed fp32 workspace
    long T, int N, int Kin, int tiles_n, int ksplit) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  bf16* a_s = reinterpret_cast<bf16*>(smem_raw);       // [K_CHUNK][LDS_N] x2
  bf16* b_s = a_s + 2 * K_CHUNK * LDS_N;               // [K_CHUNK][LDS_N] x2
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

  int buf = 0;
  stage_nt(dy, N, m0, N, kbeg, kend, a_s);
  stage_nt(x, Kin, n0, Kin, kbeg, kend, b_s);
  __syncthreads();
  for (long k0 = kbeg; k0 < kend; k0 += K_CHUNK) {
    const int cur = buf;
    if (k0 + K_CHUNK < kend) {
      const int nxt = 1 - buf;
      stage_nt(dy, N, m0, N, k0 + K_CHUNK, kend, a_s + nxt * K_CHUNK * LDS_N);
      stage_nt(x, Kin, n0, Kin, k0 + K_CHUNK, kend,
               b_s + nxt * K_CHUNK * LDS_N);
      buf = nxt;
    }
    const bf16* a = a_s + cur * K_CHUNK * LDS_N;
    const bf16* b = b_s + cur * K_CHUNK * LDS_N;
#pragma unroll
    for (int kk = 0; kk < K_CHUNK / 32; ++kk) {
      bfrag bfr[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) bfr[j] = wg_fragT(b, j * 16, kk * 32);
#pragma unroll
      for (int i = 0; i < 2; ++i) {
        const bfrag af = wg_fragT(a, (wid * 2 + i) * 16, kk * 32);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af, bfr[j], acc[i][j], 0, 0, 0);
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
  const size_t smem = (size_t)4 * K_CHUNK * LDS_N * sizeof(bf16);
  dim3 grid(tiles_m * tiles_n, 1, (unsigned)ksplit);
  hipLaunchKernelGGL(wgrad_kernel, grid, dim3(256), smem, cur_stream(dy),
                     (const bf16*)dy.data_ptr(), (const bf16*)x.data_ptr(),
                     ws.data_ptr<float>(), T, N, Kin, tiles_n, ksplit);
  HIP_CHECK_LAST();
  return ws;
}

// ---------------------------------------------------------------------
// v3 (EXPERIMENTAL, exposed as wgrad2): same tiling as v2, but the B
// (= X) tile is transposed inside LDS in a second staging pass so the 8
// B-fragments per kk step become single ds_read_b128s instead of 8
// scalar reads each. B-frag elements are read 4x (once per wave) while
// the transpose pays the scalar reads only once, so the hot-loop LDS
// instruction count drops ~3x vs v2 (80 -> ~24 read instrs per wave per
// kk). A stays in [K][N] orientation (its elements are read once, so a
// transpose would not pay). bT row stride 72 elements = 144 B keeps the
// 16-lane fragment reads on 64 distinct banks.
// ---------------------------------------------------------------------
#define LDS_KT (K_CHUNK + 8)   // bT row stride (odd multiple of 16 B)

// vectorized fragment from a [N][K]-stored tile (identical indexing to
// an A-fragment: lane l -> bT[n0 + (l&15)][k0 + (l>>4)*8 .. +8])
__device__ __forceinline__ bfrag wg_fragV(const bf16* tile, int n0, int k0) {
  const int l = threadIdx.x & (WAVE - 1);
  const bf16* p = tile + (long)(n0 + (l & 15)) * LDS_KT + k0 + ((l >> 4) << 3);
  bfrag f;
#pragma unroll
  for (int e = 0; e < 8; ++e) reinterpret_cast<bf16*>(&f)[e] = p[e];
  return f;
}

// transpose raw [K_CHUNK][LDS_N] -> bT [WG_TILE][LDS_KT]; each thread
// builds whole 8-element k-vectors (8 scalar reads, one b128 write).
__device__ __forceinline__ void transpose_b(const bf16* raw, bf16* bT) {
  const int vecs = WG_TILE * (K_CHUNK / 8);          // 1024
  for (int i = threadIdx.x; i < vecs; i += blockDim.x) {
    const int k0 = (i / WG_TILE) * 8;
    const int n = i % WG_TILE;                        // lanes -> contig n
    s16x8 v;
#pragma unroll
    for (int e = 0; e < 8; ++e)
      reinterpret_cast<bf16*>(&v)[e] = raw[(long)(k0 + e) * LDS_N + n];
    *reinterpret_cast<s16x8*>(bT + (long)n * LDS_KT + k0) = v;
  }
}

__global__ __launch_bounds__(256) void wgrad2_kernel(
    const bf16* __restrict__ dy,  // [T, N]
    const bf16* __restrict__ x,   // [T, Kin]
    float* __restrict__ ws,       // [N, Kin] zeroed fp32 workspace
    long T, int N, int Kin, int tiles_n, int ksplit) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  bf16* a_s = reinterpret_cast<bf16*>(smem_raw);       // [K_CHUNK][LDS_N] x2
  bf16* b_s = a_s + 2 * K_CHUNK * LDS_N;               // [K_CHUNK][LDS_N] x2
  bf16* bt_s = b_s + 2 * K_CHUNK * LDS_N;              // [WG_TILE][LDS_KT] x2
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

  int buf = 0;
  stage_nt(dy, N, m0, N, kbeg, kend, a_s);
  stage_nt(x, Kin, n0, Kin, kbeg, kend, b_s);
  __syncthreads();
  transpose_b(b_s, bt_s);
  __syncthreads();
  for (long k0 = kbeg; k0 < kend; k0 += K_CHUNK) {
    const int cur = buf;
    if (k0 + K_CHUNK < kend) {
      const int nxt = 1 - buf;
      stage_nt(dy, N, m0, N, k0 + K_CHUNK, kend, a_s + nxt * K_CHUNK * LDS_N);
      stage_nt(x, Kin, n0, Kin, k0 + K_CHUNK, kend,
               b_s + nxt * K_CHUNK * LDS_N);
      buf = nxt;
    }
    const bf16* a = a_s + cur * K_CHUNK * LDS_N;
    const bf16* bT = bt_s + cur * WG_TILE * LDS_KT;
#pragma unroll
    for (int kk = 0; kk < K_CHUNK / 32; ++kk) {
      bfrag bfr[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) bfr[j] = wg_fragV(bT, j * 16, kk * 32);
#pragma unroll
      for (int i = 0; i < 2; ++i) {
        const bfrag af = wg_fragT(a, (wid * 2 + i) * 16, kk * 32);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af, bfr[j], acc[i][j], 0, 0, 0);
      }
    }
    __syncthreads();
    if (k0 + K_CHUNK < kend) {
      // raw B slab for the NEXT chunk just landed in b_s[buf]
      transpose_b(b_s + buf * K_CHUNK * LDS_N, bt_s + buf * WG_TILE * LDS_KT);
      __syncthreads();
    }
  }

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

at::Tensor wgrad2(const at::Tensor& dy, const at::Tensor& x, long ksplit) {
  CHECK_CUDA_CONTIG(dy);
  CHECK_CUDA_CONTIG(x);
  TORCH_CHECK(dy.scalar_type() == at::kBFloat16 &&
                  x.scalar_type() == at::kBFloat16,
              "wgrad2: bf16 operands");
  const long T = dy.size(0);
  const int N = dy.size(1);
  const int Kin = x.size(1);
  TORCH_CHECK(x.size(0) == T, "wgrad2: token dims differ");
  TORCH_CHECK(N % 8 == 0 && Kin % 8 == 0, "wgrad2: dims % 8");
  auto ws = at::zeros({(long)N, (long)Kin},
                      dy.options().dtype(at::kFloat));
  const int tiles_m = (N + WG_TILE - 1) / WG_TILE;
  const int tiles_n = (Kin + WG_TILE - 1) / WG_TILE;
  if (ksplit <= 0) {
    const long want = 512;
    ksplit = std::max<long>(1, want / std::max(1, tiles_m * tiles_n));
    ksplit = std::min<long>(ksplit, (T + K_CHUNK - 1) / K_CHUNK);
  }
  const size_t smem = (size_t)(4 * K_CHUNK * LDS_N
                               + 2 * WG_TILE * LDS_KT) * sizeof(bf16);
  dim3 grid(tiles_m * tiles_n, 1, (unsigned)ksplit);
  hipLaunchKernelGGL(wgrad2_kernel, grid, dim3(256), smem, cur_stream(dy),
                     (const bf16*)dy.data_ptr(), (const bf16*)x.data_ptr(),
                     ws.data_ptr<float>(), T, N, Kin, tiles_n, ksplit);
  HIP_CHECK_LAST();
  return ws;
}
