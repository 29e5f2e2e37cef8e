// MFMA fragment-layout probe: packs A[16][32], B-as-[N][K] per the
// layout assumed across this codebase, runs one mfma_f32_16x16x32_bf16,
// unpacks C per the assumed C layout. Host test compares against A@B —
// a mismatch localizes which of the three mappings is wrong.
#include "common.h"

__global__ void mfma_probe_kernel(const float* __restrict__ a,   // [16][32]
                                  const float* __restrict__ bt,  // [16][32] = B^T
                                  float* __restrict__ c) {       // [16][16]
  const int l = threadIdx.x & (WAVE - 1);
  mfma_bf16x8 af, bf;
#pragma unroll
  for (int e = 0; e < 8; ++e) {
    const int row = l & 15;
    const int k = ((l >> 4) << 3) + e;
    reinterpret_cast<bf16*>(&af)[e] = __float2bfloat16(a[row * 32 + k]);
    reinterpret_cast<bf16*>(&bf)[e] = __float2bfloat16(bt[row * 32 + k]);
  }
  mfma_f32x4 acc{0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf, acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int row = ((l >> 4) << 2) + r;
    const int col = l & 15;
    c[row * 16 + col] = acc[r];
  }
}

at::Tensor mfma_probe(const at::Tensor& a, const at::Tensor& bt) {
  CHECK_CUDA_CONTIG(a);
  auto c = at::zeros({16, 16}, a.options());
  hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, cur_stream(a),
                     a.data_ptr<float>(), bt.data_ptr<float>(),
                     c.data_ptr<float>());
  HIP_CHECK_LAST();
  return c;
}
