// Shared helpers for the gfx950 (CDNA4) kernels.
// Hardware model per /opt/skills/guides: wave = 64 lanes, 4 SIMD-32/CU,
// LDS 160 KiB/CU, MFMA bf16 16x16x32.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#define WAVE 64

#define HIP_CHECK_LAST()                                                     \
  do {                                                                       \
    hipError_t e = hipGetLastError();                                        \
    TORCH_CHECK(e == hipSuccess, "HIP kernel launch failed: ",               \
                hipGetErrorString(e));                                       \
  } while (0)

#define CHECK_CUDA_CONTIG(t)                                                 \
  TORCH_CHECK((t).is_cuda() && (t).is_contiguous(), #t                       \
              " must be a contiguous GPU tensor")

static inline hipStream_t cur_stream(const at::Tensor& t) {
  return c10::hip::getCurrentHIPStream(t.get_device()).stream();
}

// ---------------------------------------------------------------- dtypes
using bf16 = __hip_bfloat16;

template <typename T> struct AccType { using type = float; };

__device__ __forceinline__ float to_f32(float x) { return x; }
__device__ __forceinline__ float to_f32(bf16 x) { return __bfloat162float(x); }
__device__ __forceinline__ void from_f32(float v, float* o) { *o = v; }
__device__ __forceinline__ void from_f32(float v, bf16* o) {
  *o = __float2bfloat16(v);
}

// vector types for wide loads (guideline 13: always vectorize bf16)
typedef __attribute__((ext_vector_type(2))) float   f32x2;
typedef __attribute__((ext_vector_type(4))) float   f32x4;
typedef __attribute__((ext_vector_type(4))) short   s16x4;
typedef __attribute__((ext_vector_type(8))) short   s16x8;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;

// MFMA fragment types (16x16x32 bf16: A/B 8 bf16 = 4 VGPR, C/D 4 f32)
typedef __attribute__((ext_vector_type(4))) float mfma_f32x4;
typedef __attribute__((ext_vector_type(8))) __bf16 mfma_bf16x8;

// -------------------------------------------------------- wave reductions
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off);
  return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off));
  return v;
}

// reduce across a 16-lane group (lanes sharing lane>>4), used for MFMA
// C-layout row reductions (col = lane & 15)
__device__ __forceinline__ float group16_reduce_sum(float v) {
#pragma unroll
  for (int off = 8; off > 0; off >>= 1) v += __shfl_xor(v, off);
  return v;
}

__device__ __forceinline__ float group16_reduce_max(float v) {
#pragma unroll
  for (int off = 8; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off));
  return v;
}

// block reduction via LDS (blockDim.x threads, caller provides smem of
// blockDim.x/WAVE floats)
__device__ __forceinline__ float block_reduce_sum(float v, float* smem) {
  int lane = threadIdx.x & (WAVE - 1);
  int wid = threadIdx.x / WAVE;
  v = wave_reduce_sum(v);
  if (lane == 0) smem[wid] = v;
  __syncthreads();
  int nw = blockDim.x / WAVE;
  v = (threadIdx.x < nw) ? smem[threadIdx.x] : 0.f;
  if (wid == 0) {
#pragma unroll
    for (int off = 8; off > 0; off >>= 1) v += __shfl_xor(v, off);
  }
  if (threadIdx.x == 0) smem[0] = v;
  __syncthreads();
  return smem[0];
}
