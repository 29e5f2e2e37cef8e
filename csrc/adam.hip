// Multi-tensor AdamWeightDecay: one launch updates every
// parameter in a group. BERT AdamWeightDecay math (no bias correction,
// decoupled weight decay — reference tools/train_utils.py:276-284):
//   m = b1*m + (1-b1)*g ; v = b2*v + (1-b2)*g^2
//   p -= lr * (m / (sqrt(v)+eps) + wd*p)
//
// Two parameter modes:
//  - fp32 params: p/g/m/v all fp32 (CPU-parity path)
//  - bf16 params: model weights + grads bf16, fp32 MASTER copy updated in
//    fp32 and rounded back to the bf16 weight — the MI355X-native "pure
//    bf16" training mode (no autocast weight-cast kernels per step).
// Plus a fused multi-tensor L2 norm for the global-norm gradient clip.
#include "common.h"

#define MAX_TENSORS 512

struct ChunkMeta {
  void* p;        // bf16 or fp32 weight
  const void* g;  // grad, same dtype as p
  float* master;  // fp32 master (nullptr when p is fp32)
  float* m;
  float* v;
  long n;
  float lr;
  float wd;
};

template <typename T>
__device__ __forceinline__ void adamw_one(const ChunkMeta& mt, long i,
                                          float b1, float b2, float eps,
                                          T* p, const T* g, float mul) {
  const float gv = to_f32(g[i]);
  const float m = b1 * mt.m[i] + (1.f - b1) * gv;
  const float v = b2 * mt.v[i] + (1.f - b2) * gv * gv;
  mt.m[i] = m;
  mt.v[i] = v;
  const float pv = mt.master ? mt.master[i] : to_f32(p[i]);
  const float upd =
      pv - mul * mt.lr * (m * __frcp_rn(sqrtf(v) + eps) + mt.wd * pv);
  if (mt.master) mt.master[i] = upd;
  from_f32(upd, &p[i]);
}

// lr_mul: optional device scalar multiplied into every tensor's lr —
// lets a hipGraph-captured step keep a live LR schedule (the schedule
// writes the device scalar outside the graph; meta.lr holds the
// per-group scale).
template <typename T>
__global__ void multi_tensor_adamw_kernel(ChunkMeta* metas, int n_tensors,
                                          float b1, float b2, float eps,
                                          const float* lr_mul) {
  const float mul = lr_mul ? *lr_mul : 1.f;
  // explicit 4-wide vector I/O on every stream (g/p via 4x16b or f32x4,
  // m/v/master via f32x4) — scalar per-lane math in registers
  const long stride = (long)gridDim.x * blockDim.x * 4;
  for (int ti = blockIdx.y; ti < n_tensors; ti += gridDim.y) {
    const ChunkMeta mt = metas[ti];
    T* p = reinterpret_cast<T*>(mt.p);
    const T* g = reinterpret_cast<const T*>(mt.g);
    const long base = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 4;
    long i = base;
    for (; i + 3 < mt.n; i += stride) {
      float gv[4], pv[4];
      if (sizeof(T) == 2) {
        const s16x4 g4 = *reinterpret_cast<const s16x4*>(g + i);
        const bf16* gb = reinterpret_cast<const bf16*>(&g4);
#pragma unroll
        for (int e = 0; e < 4; ++e) gv[e] = to_f32(gb[e]);
      } else {
        const f32x4 g4 = *reinterpret_cast<const f32x4*>(
            reinterpret_cast<const float*>(g) + i);
#pragma unroll
        for (int e = 0; e < 4; ++e) gv[e] = g4[e];
      }
      f32x4 m4 = *reinterpret_cast<const f32x4*>(mt.m + i);
      f32x4 v4 = *reinterpret_cast<const f32x4*>(mt.v + i);
      if (mt.master) {
        const f32x4 ma = *reinterpret_cast<const f32x4*>(mt.master + i);
#pragma unroll
        for (int e = 0; e < 4; ++e) pv[e] = ma[e];
      } else {
#pragma unroll
        for (int e = 0; e < 4; ++e)
          pv[e] = sizeof(T) == 2
                      ? to_f32(p[i + e])
                      : reinterpret_cast<const float*>(p)[i + e];
      }
      f32x4 up;
#pragma unroll
      for (int e = 0; e < 4; ++e) {
        const float m = b1 * m4[e] + (1.f - b1) * gv[e];
        const float v = b2 * v4[e] + (1.f - b2) * gv[e] * gv[e];
        m4[e] = m;
        v4[e] = v;
        up[e] = pv[e] - mul * mt.lr *
                    (m * __frcp_rn(sqrtf(v) + eps) + mt.wd * pv[e]);
      }
      *reinterpret_cast<f32x4*>(mt.m + i) = m4;
      *reinterpret_cast<f32x4*>(mt.v + i) = v4;
      if (mt.master) *reinterpret_cast<f32x4*>(mt.master + i) = up;
      if (sizeof(T) == 2) {
        s16x4 p4;
        bf16* pb = reinterpret_cast<bf16*>(&p4);
#pragma unroll
        for (int e = 0; e < 4; ++e) pb[e] = __float2bfloat16(up[e]);
        *reinterpret_cast<s16x4*>(p + i) = p4;
      } else {
        *reinterpret_cast<f32x4*>(reinterpret_cast<float*>(p) + i) = up;
      }
    }
    for (; i < mt.n; ++i) adamw_one<T>(mt, i, b1, b2, eps, p, g, mul);
  }
}

static void launch_adamw(std::vector<ChunkMeta>& metas, long total,
                         bool is_bf16, const at::Tensor& ref, double b1,
                         double b2, double eps, const float* lr_mul) {
  const int n = metas.size();
  auto meta_blob = at::from_blob(metas.data(), {(long)(n * sizeof(ChunkMeta))},
                                 at::TensorOptions().dtype(at::kByte))
                       .to(ref.device(), /*non_blocking=*/false);
  const int block = 256;
  const int gx = std::min<long>((total / n + block - 1) / block, 512);
  dim3 grid(std::max(gx, 1), std::min(n, 64));
  auto stream = cur_stream(ref);
  if (is_bf16) {
    hipLaunchKernelGGL(multi_tensor_adamw_kernel<bf16>, grid, dim3(block), 0,
                       stream, (ChunkMeta*)meta_blob.data_ptr(), n, (float)b1,
                       (float)b2, (float)eps, lr_mul);
  } else {
    hipLaunchKernelGGL(multi_tensor_adamw_kernel<float>, grid, dim3(block), 0,
                       stream, (ChunkMeta*)meta_blob.data_ptr(), n, (float)b1,
                       (float)b2, (float)eps, lr_mul);
  }
  HIP_CHECK_LAST();
}

// Build the device-side ChunkMeta blob once (H2D copy here) so a
// hipGraph-captured step can run the kernel without any host transfer.
// Returns {blob, [total, n, is_bf16]}.
std::vector<at::Tensor> adamw_build_meta(
    std::vector<at::Tensor> params, std::vector<at::Tensor> grads,
    std::vector<at::Tensor> masters, std::vector<at::Tensor> ms,
    std::vector<at::Tensor> vs, std::vector<double> lrs,
    std::vector<double> wds) {
  const int n = params.size();
  TORCH_CHECK(n > 0 && n <= MAX_TENSORS, "adamw_build_meta: bad count");
  const bool has_master = !masters.empty();
  const bool is_bf16 = params[0].scalar_type() == at::kBFloat16;
  std::vector<ChunkMeta> metas(n);
  long total = 0;
  for (int i = 0; i < n; ++i) {
    metas[i] = {params[i].data_ptr(), grads[i].data_ptr(),
                has_master ? masters[i].data_ptr<float>() : nullptr,
                ms[i].data_ptr<float>(), vs[i].data_ptr<float>(),
                params[i].numel(), (float)lrs[i], (float)wds[i]};
    total += metas[i].n;
  }
  auto blob = at::from_blob(metas.data(), {(long)(n * sizeof(ChunkMeta))},
                            at::TensorOptions().dtype(at::kByte))
                  .to(params[0].device(), false);
  auto info = at::tensor({total, (long)n, (long)(is_bf16 ? 1 : 0)},
                         at::TensorOptions().dtype(at::kLong));
  return {blob, info};
}

// Capture-safe launch from a prebuilt blob (no host transfers).
void multi_tensor_adamw_run(const at::Tensor& blob, long total, long n,
                            bool is_bf16, double b1, double b2, double eps,
                            c10::optional<at::Tensor> lr_mul) {
  const int block = 256;
  const int gx = std::min<long>((total / n + block - 1) / block, 512);
  dim3 grid(std::max(gx, 1), std::min<long>(n, 64));
  auto stream = cur_stream(blob);
  const float* mul = nullptr;
  if (lr_mul.has_value()) mul = lr_mul->data_ptr<float>();
  if (is_bf16) {
    hipLaunchKernelGGL(multi_tensor_adamw_kernel<bf16>, grid, dim3(block), 0,
                       stream, (ChunkMeta*)blob.data_ptr(), (int)n, (float)b1,
                       (float)b2, (float)eps, mul);
  } else {
    hipLaunchKernelGGL(multi_tensor_adamw_kernel<float>, grid, dim3(block), 0,
                       stream, (ChunkMeta*)blob.data_ptr(), (int)n, (float)b1,
                       (float)b2, (float)eps, mul);
  }
  HIP_CHECK_LAST();
}

// fp32 path (masters empty) or bf16 path (masters[i] fp32, same numel).
void multi_tensor_adamw(std::vector<at::Tensor> params,
                        std::vector<at::Tensor> grads,
                        std::vector<at::Tensor> masters,
                        std::vector<at::Tensor> ms, std::vector<at::Tensor> vs,
                        std::vector<double> lrs, std::vector<double> wds,
                        double b1, double b2, double eps,
                        c10::optional<at::Tensor> lr_mul) {
  const int n = params.size();
  TORCH_CHECK(n > 0 && n <= MAX_TENSORS, "multi_tensor_adamw: bad tensor count");
  const bool has_master = !masters.empty();
  const bool is_bf16 = params[0].scalar_type() == at::kBFloat16;
  TORCH_CHECK(!is_bf16 || has_master,
              "adamw: bf16 params need fp32 master weights");
  std::vector<ChunkMeta> metas(n);
  long total = 0;
  for (int i = 0; i < n; ++i) {
    TORCH_CHECK(params[i].scalar_type() ==
                    (is_bf16 ? at::kBFloat16 : at::kFloat),
                "adamw: mixed param dtypes in one call");
    TORCH_CHECK(grads[i].is_contiguous() &&
                    grads[i].scalar_type() == params[i].scalar_type(),
                "adamw: grad must be contiguous, same dtype as param");
    metas[i] = {params[i].data_ptr(), grads[i].data_ptr(),
                has_master ? masters[i].data_ptr<float>() : nullptr,
                ms[i].data_ptr<float>(), vs[i].data_ptr<float>(),
                params[i].numel(), (float)lrs[i], (float)wds[i]};
    total += metas[i].n;
  }
  const float* mul = nullptr;
  if (lr_mul.has_value()) {
    TORCH_CHECK(lr_mul->scalar_type() == at::kFloat && lr_mul->numel() == 1,
                "lr_mul must be a 1-element fp32 device tensor");
    mul = lr_mul->data_ptr<float>();
  }
  launch_adamw(metas, total, is_bf16, params[0], b1, b2, eps, mul);
}

// ------------------------------------------------------- global L2 norm
struct NormMeta {
  const void* g;
  long n;
  int is_bf16;
};

__global__ void multi_tensor_l2_kernel(NormMeta* metas, int n_tensors,
                                       float* out) {
  // every block walks every tensor (grid-stride over elements, 4-wide
  // vector loads); block-level LDS reduce -> ONE atomic per block
  __shared__ float warp_acc[16];
  float acc = 0.f;
  const long stride = (long)gridDim.x * blockDim.x * 4;
  for (int ti = 0; ti < n_tensors; ++ti) {
    const NormMeta mt = metas[ti];
    const long base = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 4;
    if (mt.is_bf16) {
      const bf16* g = reinterpret_cast<const bf16*>(mt.g);
      long i = base;
      for (; i + 3 < mt.n; i += stride) {
        const s16x4 v4 = *reinterpret_cast<const s16x4*>(g + i);
        const bf16* vb = reinterpret_cast<const bf16*>(&v4);
#pragma unroll
        for (int e = 0; e < 4; ++e) {
          const float v = to_f32(vb[e]);
          acc += v * v;
        }
      }
      for (; i < mt.n; ++i) {  // ragged tail (rare: params are 4-aligned)
        const float v = to_f32(g[i]);
        acc += v * v;
      }
    } else {
      const float* g = reinterpret_cast<const float*>(mt.g);
      long i = base;
      for (; i + 3 < mt.n; i += stride) {
        const f32x4 v4 = *reinterpret_cast<const f32x4*>(g + i);
#pragma unroll
        for (int e = 0; e < 4; ++e) acc += v4[e] * v4[e];
      }
      for (; i < mt.n; ++i) acc += g[i] * g[i];
    }
  }
  for (int off = WAVE / 2; off > 0; off >>= 1)
    acc += __shfl_down(acc, off, WAVE);
  const int wid = threadIdx.x / WAVE;
  if ((threadIdx.x & (WAVE - 1)) == 0) warp_acc[wid] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    float s = 0.f;
    for (int w = 0; w < (int)(blockDim.x / WAVE); ++w) s += warp_acc[w];
    atomicAdd(out, s);
  }
}

std::vector<at::Tensor> norm_build_meta(std::vector<at::Tensor> grads) {
  const int n = grads.size();
  TORCH_CHECK(n > 0 && n <= MAX_TENSORS, "norm_build_meta: bad count");
  std::vector<NormMeta> metas(n);
  for (int i = 0; i < n; ++i)
    metas[i] = {grads[i].data_ptr(), grads[i].numel(),
                grads[i].scalar_type() == at::kBFloat16 ? 1 : 0};
  auto blob = at::from_blob(metas.data(), {(long)(n * sizeof(NormMeta))},
                            at::TensorOptions().dtype(at::kByte))
                  .to(grads[0].device(), false);
  auto info = at::tensor({(long)n}, at::TensorOptions().dtype(at::kLong));
  return {blob, info};
}

at::Tensor multi_tensor_sumsq_run(const at::Tensor& blob, long n,
                                  const at::Tensor& out) {
  hipLaunchKernelGGL(multi_tensor_l2_kernel, dim3(512, 1), dim3(256), 0,
                     cur_stream(blob), (NormMeta*)blob.data_ptr(), (int)n,
                     out.data_ptr<float>());
  HIP_CHECK_LAST();
  return out;
}


// Returns a 1-element fp32 tensor holding sum of squares (caller sqrts).
at::Tensor multi_tensor_sumsq(std::vector<at::Tensor> grads) {
  const int n = grads.size();
  TORCH_CHECK(n > 0 && n <= MAX_TENSORS, "multi_tensor_sumsq: bad count");
  std::vector<NormMeta> metas(n);
  long total = 0;
  for (int i = 0; i < n; ++i) {
    TORCH_CHECK(grads[i].is_contiguous(), "sumsq: non-contiguous grad");
    metas[i] = {grads[i].data_ptr(), grads[i].numel(),
                grads[i].scalar_type() == at::kBFloat16 ? 1 : 0};
    total += metas[i].n;
  }
  auto out = at::zeros({1}, grads[0].options().dtype(at::kFloat));
  auto meta_blob = at::from_blob(metas.data(), {(long)(n * sizeof(NormMeta))},
                                 at::TensorOptions().dtype(at::kByte))
                       .to(grads[0].device(), false);
  const int block = 256;
  // ~2048 waves total saturate HBM for a read-only reduction
  dim3 grid(512, 1);
  hipLaunchKernelGGL(multi_tensor_l2_kernel, grid, dim3(block), 0,
                     cur_stream(grads[0]), (NormMeta*)meta_blob.data_ptr(), n,
                     out.data_ptr<float>());
  HIP_CHECK_LAST();
  return out;
}

// Scale every grad in-place by *scale (1-element fp32 tensor, device side —
// no host sync in the clip path).
struct ScaleMeta {
  void* g;
  long n;
  int is_bf16;
};

__global__ void multi_tensor_scale_kernel(ScaleMeta* metas, int n_tensors,
                                          const float* scale) {
  const float s = *scale;
  const long stride = (long)gridDim.x * blockDim.x * 4;
  for (int ti = blockIdx.y; ti < n_tensors; ti += gridDim.y) {
    const ScaleMeta mt = metas[ti];
    const long base = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 4;
    if (mt.is_bf16) {
      bf16* g = reinterpret_cast<bf16*>(mt.g);
      long i = base;
      for (; i + 3 < mt.n; i += stride) {
        s16x4 v4 = *reinterpret_cast<const s16x4*>(g + i);
        bf16* vb = reinterpret_cast<bf16*>(&v4);
#pragma unroll
        for (int e = 0; e < 4; ++e)
          vb[e] = __float2bfloat16(to_f32(vb[e]) * s);
        *reinterpret_cast<s16x4*>(g + i) = v4;
      }
      for (; i < mt.n; ++i) g[i] = __float2bfloat16(to_f32(g[i]) * s);
    } else {
      float* g = reinterpret_cast<float*>(mt.g);
      long i = base;
      for (; i + 3 < mt.n; i += stride) {
        f32x4 v4 = *reinterpret_cast<const f32x4*>(g + i);
#pragma unroll
        for (int e = 0; e < 4; ++e) v4[e] *= s;
        *reinterpret_cast<f32x4*>(g + i) = v4;
      }
      for (; i < mt.n; ++i) g[i] *= s;
    }
  }
}

void multi_tensor_scale(std::vector<at::Tensor> grads,
                        const at::Tensor& scale) {
  const int n = grads.size();
  TORCH_CHECK(n > 0 && n <= MAX_TENSORS, "multi_tensor_scale: bad count");
  std::vector<ScaleMeta> metas(n);
  long total = 0;
  for (int i = 0; i < n; ++i) {
    metas[i] = {grads[i].data_ptr(), grads[i].numel(),
                grads[i].scalar_type() == at::kBFloat16 ? 1 : 0};
    total += metas[i].n;
  }
  auto meta_blob = at::from_blob(metas.data(), {(long)(n * sizeof(ScaleMeta))},
                                 at::TensorOptions().dtype(at::kByte))
                       .to(grads[0].device(), false);
  const int block = 256;
  const int gx = std::min<long>((total / n + block - 1) / block, 256);
  dim3 grid(std::max(gx, 1), std::min(n, 64));
  hipLaunchKernelGGL(multi_tensor_scale_kernel, grid, dim3(block), 0,
                     cur_stream(grads[0]), (ScaleMeta*)meta_blob.data_ptr(), n,
                     scale.data_ptr<float>());
  HIP_CHECK_LAST();
}

void multi_tensor_scale_run(const at::Tensor& blob, long n,
                            const at::Tensor& scale) {
  // NormMeta and ScaleMeta share layout {ptr, n, is_bf16}
  hipLaunchKernelGGL(multi_tensor_scale_kernel, dim3(256, 64), dim3(256), 0,
                     cur_stream(blob), (ScaleMeta*)blob.data_ptr(), (int)n,
                     scale.data_ptr<float>());
  HIP_CHECK_LAST();
}
