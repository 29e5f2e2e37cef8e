// Multi-tensor AdamWeightDecay: one launch updates every
// parameter in a group. BERT AdamWeightDecay math (no bias correction,
// decoupled weight decay — reference tools/train_utils.py:276-284):
//   m = b1*m + (1-b1)*g ; v = b2*v + (1-b2)*g^2
//   p -= lr * (m / (sqrt(v)+eps) + wd*p)
#include "common.h"

#define MAX_TENSORS 512

struct ChunkMeta {
  float* p;
  float* g;
  float* m;
  float* v;
  long n;
  float lr;
  float wd;
};

__global__ void multi_tensor_adamw_kernel(ChunkMeta* metas, int n_tensors,
                                          float b1, float b2, float eps) {
  // grid-stride over (tensor, element) pairs: block handles slices of one
  // tensor chosen by blockIdx.y-style flattening
  for (int ti = blockIdx.y; ti < n_tensors; ti += gridDim.y) {
    const ChunkMeta mt = metas[ti];
    const float lr = mt.lr;
    const float wd = mt.wd;
    const long stride = (long)gridDim.x * blockDim.x;
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < mt.n;
         i += stride) {
      const float g = mt.g[i];
      const float m = b1 * mt.m[i] + (1.f - b1) * g;
      const float v = b2 * mt.v[i] + (1.f - b2) * g * g;
      mt.m[i] = m;
      mt.v[i] = v;
      mt.p[i] -= lr * (m * __frcp_rn(sqrtf(v) + eps) + wd * mt.p[i]);
    }
  }
}

void multi_tensor_adamw(std::vector<at::Tensor> params,
                        std::vector<at::Tensor> grads,
                        std::vector<at::Tensor> ms, std::vector<at::Tensor> vs,
                        std::vector<double> lrs, std::vector<double> wds,
                        double b1, double b2, double eps) {
  const int n = params.size();
  TORCH_CHECK(n > 0 && n <= MAX_TENSORS, "multi_tensor_adamw: bad tensor count");
  std::vector<ChunkMeta> metas(n);
  long total = 0;
  for (int i = 0; i < n; ++i) {
    TORCH_CHECK(params[i].scalar_type() == at::kFloat,
                "adamw: fp32 master params expected");
    TORCH_CHECK(grads[i].is_contiguous(), "adamw: non-contiguous grad");
    metas[i] = {params[i].data_ptr<float>(), grads[i].data_ptr<float>(),
                ms[i].data_ptr<float>(), vs[i].data_ptr<float>(),
                params[i].numel(), (float)lrs[i], (float)wds[i]};
    total += metas[i].n;
  }
  auto meta_blob = at::from_blob(metas.data(), {(long)(n * sizeof(ChunkMeta))},
                                 at::TensorOptions().dtype(at::kByte))
                       .to(params[0].device(), /*non_blocking=*/false);
  const int block = 256;
  const int gx = std::min<long>((total / n + block - 1) / block, 512);
  dim3 grid(std::max(gx, 1), std::min(n, 64));
  hipLaunchKernelGGL(multi_tensor_adamw_kernel, grid, dim3(block), 0,
                     cur_stream(params[0]),
                     (ChunkMeta*)meta_blob.data_ptr(), n, (float)b1,
                     (float)b2, (float)eps);
  HIP_CHECK_LAST();
}
