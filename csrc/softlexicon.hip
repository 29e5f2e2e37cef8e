// SoftLexicon gather-scale-reduce (the word-enhance fusion
// kernel named in BASELINE.json): table [V,E] fp32, ids/weights [N,40]
// (4 roles x 10 slots per token) -> out [N,4E].
// One wave per token: slot ids/weights broadcast via shfl, lanes own
// embedding columns; fp32 accumulate.
#include "common.h"

#define ROLES 4
#define SLOTS 10

__global__ void softlexicon_fwd_kernel(const float* __restrict__ table,
                                       const int* __restrict__ ids,
                                       const float* __restrict__ weights,
                                       float* __restrict__ out,
                                       long N, int V, int E) {
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const long tok = (long)blockIdx.x * (blockDim.x / WAVE) + wid;
  if (tok >= N) return;
  const int* idr = ids + tok * ROLES * SLOTS;
  const float* wr = weights + tok * ROLES * SLOTS;
  // lane holds slot metadata for slot = lane (40 slots < 64 lanes)
  const int my_id = (lane < ROLES * SLOTS) ? idr[lane] : 0;
  const float my_w = (lane < ROLES * SLOTS) ? wr[lane] : 0.f;
  for (int r = 0; r < ROLES; ++r) {
    float acc = 0.f;  // lane -> column e (E <= 64)
    for (int s = 0; s < SLOTS; ++s) {
      const int id = __shfl(my_id, r * SLOTS + s);
      const float w = __shfl(my_w, r * SLOTS + s);
      if (lane < E && w != 0.f) acc += table[(long)id * E + lane] * w;
    }
    if (lane < E) out[tok * ROLES * E + r * E + lane] = acc;
  }
}

__global__ void softlexicon_bwd_kernel(const float* __restrict__ dout,
                                       const float* __restrict__ table,
                                       const int* __restrict__ ids,
                                       const float* __restrict__ weights,
                                       float* __restrict__ dtable,   // zeroed
                                       float* __restrict__ dweights, // [N,40]
                                       long N, int V, int E) {
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const long tok = (long)blockIdx.x * (blockDim.x / WAVE) + wid;
  if (tok >= N) return;
  const int* idr = ids + tok * ROLES * SLOTS;
  const float* wr = weights + tok * ROLES * SLOTS;
  const int my_id = (lane < ROLES * SLOTS) ? idr[lane] : 0;
  const float my_w = (lane < ROLES * SLOTS) ? wr[lane] : 0.f;
  for (int r = 0; r < ROLES; ++r) {
    const float g = (lane < E) ? dout[tok * ROLES * E + r * E + lane] : 0.f;
    for (int s = 0; s < SLOTS; ++s) {
      const int k = r * SLOTS + s;
      const int id = __shfl(my_id, k);
      const float w = __shfl(my_w, k);
      // dweights[k] = dot(dout_role, table[id])
      float dw = (lane < E) ? g * table[(long)id * E + lane] : 0.f;
      dw = wave_reduce_sum(dw);
      if (lane == 0) dweights[tok * ROLES * SLOTS + k] = dw;
      // dtable[id] += w * dout_role  (atomic: frequent words collide)
      if (lane < E && w != 0.f) atomicAdd(&dtable[(long)id * E + lane], w * g);
    }
  }
}

std::vector<at::Tensor> softlexicon_bwd(const at::Tensor& dout,
                                        const at::Tensor& table,
                                        const at::Tensor& ids,
                                        const at::Tensor& weights) {
  CHECK_CUDA_CONTIG(dout);
  const long N = (long)ids.size(0) * ids.size(1);
  const int V = table.size(0), E = table.size(1);
  auto dtable = at::zeros_like(table);
  auto dweights = at::empty_like(weights);
  const int nw = 4;
  hipLaunchKernelGGL(softlexicon_bwd_kernel, dim3((N + nw - 1) / nw),
                     dim3(nw * WAVE), 0, cur_stream(dout),
                     dout.data_ptr<float>(), table.data_ptr<float>(),
                     ids.data_ptr<int>(), weights.data_ptr<float>(),
                     dtable.data_ptr<float>(), dweights.data_ptr<float>(),
                     N, V, E);
  HIP_CHECK_LAST();
  return {dtable, dweights};
}

at::Tensor softlexicon_fwd(const at::Tensor& table, const at::Tensor& ids,
                           const at::Tensor& weights) {
  CHECK_CUDA_CONTIG(table);
  CHECK_CUDA_CONTIG(ids);
  TORCH_CHECK(table.scalar_type() == at::kFloat, "softlexicon table must be fp32");
  TORCH_CHECK(ids.size(-1) == ROLES * SLOTS, "ids last dim must be 40");
  const int V = table.size(0), E = table.size(1);
  TORCH_CHECK(E <= WAVE, "softlexicon: word dim > 64");
  const long N = (long)ids.size(0) * ids.size(1);
  auto out = at::empty({ids.size(0), ids.size(1), (long)ROLES * E},
                       table.options());
  const int nw = 4;
  hipLaunchKernelGGL(softlexicon_fwd_kernel, dim3((N + nw - 1) / nw),
                     dim3(nw * WAVE), 0, cur_stream(table),
                     table.data_ptr<float>(), ids.data_ptr<int>(),
                     weights.data_ptr<float>(), out.data_ptr<float>(), N, V, E);
  HIP_CHECK_LAST();
  return out;
}
