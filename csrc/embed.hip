// Fused BERT embedding gather: one kernel computes
// out[b, l, :] = word[token[b,l]] + pos[l] + tok_type[seg[b,l]]
// replacing three separate gather launches + two adds (each a full
// [B, L, H] pass over HBM). Memory-bound by design: 3 row reads + 1 row
// write, vectorized b128. The LayerNorm + dropout that follow stay in
// the existing fused dropout_add_ln kernel.
#include "common.h"

__global__ __launch_bounds__(256) void embed3_kernel(
    const bf16* __restrict__ word,  // [V, H]
    const bf16* __restrict__ pos,   // [P, H]
    const bf16* __restrict__ tok,   // [S, H]
    const long* __restrict__ ids,   // [R] flattened token ids
    const long* __restrict__ segs,  // [R] flattened segment ids
    bf16* __restrict__ out,         // [R, H]
    long R, int H, int L) {
  const long row = blockIdx.x;
  if (row >= R) return;
  const long wrow = ids[row];
  const long srow = segs[row];
  const int prow = (int)(row % L);
  const bf16* w = word + wrow * H;
  const bf16* p = pos + (long)prow * H;
  const bf16* s = tok + srow * H;
  bf16* o = out + row * H;
  for (int c = threadIdx.x * 8; c < H; c += blockDim.x * 8) {
    const bf16x8 wv = *reinterpret_cast<const bf16x8*>(w + c);
    const bf16x8 pv = *reinterpret_cast<const bf16x8*>(p + c);
    const bf16x8 sv = *reinterpret_cast<const bf16x8*>(s + c);
    bf16x8 ov;
#pragma unroll
    for (int e = 0; e < 8; ++e)
      ov[e] = (__bf16)((float)wv[e] + (float)pv[e] + (float)sv[e]);
    *reinterpret_cast<bf16x8*>(o + c) = ov;
  }
}

// word/pos/tok [*, H] bf16; token_ids/segment_ids [B, L] int64 ->
// [B, L, H] bf16 (the position index is l, the row index mod L).
at::Tensor embed3_fwd(const at::Tensor& word, const at::Tensor& pos,
                      const at::Tensor& tok, const at::Tensor& token_ids,
                      const at::Tensor& segment_ids) {
  CHECK_CUDA_CONTIG(word);
  CHECK_CUDA_CONTIG(pos);
  CHECK_CUDA_CONTIG(tok);
  CHECK_CUDA_CONTIG(token_ids);
  CHECK_CUDA_CONTIG(segment_ids);
  TORCH_CHECK(word.scalar_type() == at::kBFloat16 &&
                  pos.scalar_type() == at::kBFloat16 &&
                  tok.scalar_type() == at::kBFloat16,
              "embed3: bf16 tables");
  const int H = word.size(1);
  TORCH_CHECK(H % 8 == 0, "embed3: hidden % 8");
  const long B = token_ids.size(0), L = token_ids.size(1);
  const long R = B * L;
  auto out = at::empty({B, L, (long)H}, word.options());
  hipLaunchKernelGGL(embed3_kernel, dim3((unsigned)R), dim3(256), 0,
                     cur_stream(word), (const bf16*)word.data_ptr(),
                     (const bf16*)pos.data_ptr(), (const bf16*)tok.data_ptr(),
                     token_ids.data_ptr<long>(), segment_ids.data_ptr<long>(),
                     (bf16*)out.data_ptr(), R, H, (int)L);
  HIP_CHECK_LAST();
  return out;
}
