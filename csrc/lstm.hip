// Persistent BiLSTM recurrence (the "BiLSTM tagger" kernel
// named in BASELINE.json). The x-projection (x @ [W_ih_f|W_ih_b] + b) is a
// single library GEMM done by the Python wrapper; these kernels own the
// sequential part: per step  gates = gates_x[t] + h_{t-1} @ W_hh,
// LSTM cell update, variable-length masking.
//
// Design:
//  * BOTH directions run in ONE launch (blockIdx.y = direction) so their
//    sequence loops overlap on different CUs.
//  * Strided I/O: python layout is [B,L,8h] gates (fw|bw halves) and
//    [B,L,2h] hidden — the BiLSTM concat is free.
//  * Recurrent weights: LDS-resident for the whole sequence when
//    4h*h*2B fits (h <= 128); for h in (128, 256] the MFMA B-operands
//    stream from global memory (L2-resident: W_hh is <= 512 KiB and
//    re-read every timestep). Hidden sizes must be a multiple of 32 —
//    the wrapper zero-pads (exact: padded units stay 0 through time).
//  * One workgroup (4 waves) owns a 16-row batch tile and loops time
//    in-kernel; MFMA 16x16x32 bf16 computes the [16,4h] gate panel;
//    column frag f (16 cols) is owned by wave f%4, so each lane locally
//    combines its i/f/g/o gates and carries cell state in registers.
//  * Software pipeline: step t+1's gates_x chunks are loaded into
//    registers while step t's MFMAs run (fwd).
//
// fwd stores activated gates interleaved ([D,B,L,h,4] — one f32x4 per
// (row, j) in bwd) + cell states; bwd replays in reverse producing
// pre-activation gate grads; dW_hh / dW_ih / db / dx are library GEMMs
// in the wrapper.
#include "common.h"

#define H_MAX 256
#define NQ_MAX 4  // column frags per wave: h/16/4 <= 4 for h <= 256
#define NC_MAX 8  // gates_x chunks per thread: 8h/256 <= 8

using bfrag = mfma_bf16x8;
using cfrag = mfma_f32x4;

__device__ __forceinline__ bfrag lds_frag_l(const bf16* base, int i0, int ld,
                                            int k0) {
  const int l = threadIdx.x & (WAVE - 1);
  return *reinterpret_cast<const bfrag*>(base + (long)(i0 + (l & 15)) * ld +
                                         k0 + ((l >> 4) << 3));
}

__device__ __forceinline__ float act_f(float x, bool relu) {
  return relu ? fmaxf(x, 0.f) : tanhf(x);
}
__device__ __forceinline__ float dact_from_out(float y, bool relu) {
  // derivative expressed from the activated value y = act(x)
  return relu ? (y > 0.f ? 1.f : 0.f) : (1.f - y * y);
}

// dir = blockIdx.y. gates_x rows: stride gxs, this direction's slice at
// column offset dir*4h; hs rows: stride hss, offset dir*h. cs/gates_out
// per-direction contiguous ([D,B,L,h] / [D,B,L,h,4] interleaved).
// Direction 1 (when gridDim.y == 2) scans the sequence reversed.
template <typename T>
__global__ __launch_bounds__(256) void lstm_fwd_kernel(
    const T* __restrict__ gates_x,    // [B,L,gxs]
    const bf16* __restrict__ w_hh_t,  // [D,4h,h] (transposed by wrapper)
    const int* __restrict__ lens, T* __restrict__ hs,  // [B,L,hss]
    float* __restrict__ cs,                            // [D,B,L,h]
    float* __restrict__ gates_out,                     // [D,B,L,h,4]
    int B, int L, int h, int gxs, int hss, bool rev0, bool relu,
    float cell_clip) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  // padded strides (odd 16B groups) keep the MFMA operand reads LDS
  // bank-conflict free; gx_s reads are per-lane scalars (no padding)
  const int hs_ld = h + 8;
  bf16* hb = reinterpret_cast<bf16*>(smem_raw);  // [16][hs_ld] current h
  bf16* gx_s = hb + 16 * hs_ld;                  // [16][4h] staged gates_x
  bf16* wT = gx_s + 16 * 4 * h;                  // [4h][hs_ld] iff w_lds
  const bool w_lds = h <= 128;
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int b0 = blockIdx.x * 16;
  const int dir = blockIdx.y;
  const bool reverse = dir ? true : rev0;
  const int gx_off = dir * 4 * h;
  const int hs_off = dir * h;
  const long dbase = (long)dir * B * L;  // row offset into cs/gates_out
  const bf16* wsrc = w_hh_t + (long)dir * 4 * h * h;

  int w_ld = h;  // stride of the MFMA B-operand source
  if (w_lds) {
    for (int i = threadIdx.x; i < 4 * h * h / 8; i += blockDim.x) {
      const int r = (i * 8) / h;
      const int c = (i * 8) % h;
      *reinterpret_cast<s16x8*>(wT + (long)r * hs_ld + c) =
          reinterpret_cast<const s16x8*>(wsrc)[i];
    }
    wsrc = wT;
    w_ld = hs_ld;
  }
  for (int i = threadIdx.x; i < 16 * hs_ld / 8; i += blockDim.x)
    reinterpret_cast<s16x8*>(hb)[i] = s16x8{};
  __syncthreads();

  const int NKK = h / 32;
  const int N16 = h / 16;                  // column frags per gate
  const int NQ = (N16 + 3) / 4;            // frags per wave
  // lane-owned cell state: rows r=0..3 (row = (lane>>4)*4+r), col frag
  // f = wid + 4*q, j = f*16 + (lane&15)
  float c_reg[4][NQ_MAX] = {};
  const int lrow = ((lane >> 4) << 2);  // +r
  int mylen[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int b = b0 + lrow + r;
    mylen[r] = (b < B) ? lens[b] : 0;
  }

  // software pipeline for gates_x staging (see header)
  const int NC = 8 * h / 256;            // chunks per thread
  long chunk_base[NC_MAX];
  int chunk_off[NC_MAX];
#pragma unroll
  for (int c = 0; c < NC_MAX; ++c) {
    if (c >= NC) { chunk_base[c] = -1; continue; }
    const int i = threadIdx.x + c * blockDim.x;
    const int row = (i * 8) / (4 * h);
    const int col = (i * 8) % (4 * h);
    const int b = b0 + row;
    chunk_base[c] = (b < B) ? ((long)b * L) * gxs + gx_off + col : -1;
    chunk_off[c] = i;
  }
  s16x8 pre[NC_MAX];
  auto load_chunks = [&](int t) {
#pragma unroll
    for (int c = 0; c < NC_MAX; ++c) {
      if (c >= NC) break;
      if (chunk_base[c] < 0) { pre[c] = s16x8{}; continue; }
      const long g = chunk_base[c] + (long)t * gxs;
      if (sizeof(T) == 2) {
        pre[c] = *reinterpret_cast<const s16x8*>(gates_x + g);
      } else {
        const f32x4 lo = *reinterpret_cast<const f32x4*>(gates_x + g);
        const f32x4 hi = *reinterpret_cast<const f32x4*>(gates_x + g + 4);
        bf16 packed[8];
#pragma unroll
        for (int e = 0; e < 4; ++e) {
          packed[e] = __float2bfloat16(lo[e]);
          packed[e + 4] = __float2bfloat16(hi[e]);
        }
        pre[c] = *reinterpret_cast<const s16x8*>(packed);
      }
    }
  };
  auto store_chunks = [&]() {
#pragma unroll
    for (int c = 0; c < NC_MAX; ++c)
      if (c < NC) reinterpret_cast<s16x8*>(gx_s)[chunk_off[c]] = pre[c];
  };
  load_chunks(reverse ? L - 1 : 0);
  store_chunks();

  for (int step = 0; step < L; ++step) {
    const int t = reverse ? (L - 1 - step) : step;
    if (step + 1 < L)  // issue next step's loads early (latency overlap)
      load_chunks(reverse ? (L - 2 - step) : (step + 1));
    // gates = h_prev @ W_hh  (+ gates_x added in the epilogue)
    cfrag acc[4][NQ_MAX];
#pragma unroll
    for (int g = 0; g < 4; ++g)
#pragma unroll
      for (int q = 0; q < NQ_MAX; ++q) acc[g][q] = cfrag{0.f, 0.f, 0.f, 0.f};
    for (int kk = 0; kk < NKK; ++kk) {
      const bfrag ah = lds_frag_l(hb, 0, hs_ld, kk * 32);
#pragma unroll
      for (int g = 0; g < 4; ++g)
#pragma unroll
        for (int q = 0; q < NQ_MAX; ++q) {
          const int f = wid + 4 * q;
          if (q >= NQ || f >= N16) continue;
          const int col0 = g * h + f * 16;
          acc[g][q] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              ah, lds_frag_l(wsrc, col0, w_ld, kk * 32), acc[g][q], 0, 0, 0);
        }
    }
    __syncthreads();  // hb + gx_s reads of this step done below this point
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int b = b0 + lrow + r;
      if (b >= B) continue;
      const bool valid = t < mylen[r];
      const long gbase = (dbase + (long)b * L + t) * 4 * h;
#pragma unroll
      for (int q = 0; q < NQ_MAX; ++q) {
        const int f = wid + 4 * q;
        if (q >= NQ || f >= N16) continue;
        const int j = f * 16 + (lane & 15);
        const int lr4h = (lrow + r) * 4 * h;
        float gi = acc[0][q][r] + to_f32(gx_s[lr4h + 0 * h + j]);
        float gf = acc[1][q][r] + to_f32(gx_s[lr4h + 1 * h + j]);
        float gg = acc[2][q][r] + to_f32(gx_s[lr4h + 2 * h + j]);
        float go = acc[3][q][r] + to_f32(gx_s[lr4h + 3 * h + j]);
        gi = 1.f / (1.f + __expf(-gi));
        gf = 1.f / (1.f + __expf(-gf));
        go = 1.f / (1.f + __expf(-go));
        gg = act_f(gg, relu);
        float c_new = gf * c_reg[r][q] + gi * gg;
        // TF LSTMCell cell_clip: bounds the (relu) recurrence
        if (cell_clip > 0.f)
          c_new = fminf(fmaxf(c_new, -cell_clip), cell_clip);
        const float h_new = go * act_f(c_new, relu);
        if (valid) c_reg[r][q] = c_new;
        const long obase = ((long)b * L + t) * hss + hs_off + j;
        from_f32(valid ? h_new : 0.f, &hs[obase]);
        cs[(dbase + (long)b * L + t) * h + j] = c_reg[r][q];
        const f32x4 g4 = {gi, gf, gg, go};
        *reinterpret_cast<f32x4*>(gates_out + gbase + (long)j * 4) = g4;
        hb[(lrow + r) * hs_ld + j] =
            __float2bfloat16(valid ? h_new : to_f32(hb[(lrow + r) * hs_ld + j]));
      }
    }
    __syncthreads();  // hb updated; gx_s safe to overwrite for next step
    if (step + 1 < L) store_chunks();
  }
}

template <typename T>
__global__ __launch_bounds__(256) void lstm_bwd_kernel(
    const T* __restrict__ dhs,        // [B,L,hss] upstream grad (strided)
    const float* __restrict__ cs,     // [D,B,L,h] carried cell states
    const float* __restrict__ gates,  // [D,B,L,h,4] activated, interleaved
    const bf16* __restrict__ w_hh,    // [D,h,4h] (original layout)
    const int* __restrict__ lens, T* __restrict__ dgates_x,  // [B,L,gxs]
    int B, int L, int h, int gxs, int hss, bool rev0, bool relu,
    float cell_clip) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  const int g_ld = 4 * h + 8;  // padded (bank-conflict-free frag reads)
  bf16* dg_s = reinterpret_cast<bf16*>(smem_raw);  // [16][g_ld]
  bf16* w_s = dg_s + 16 * g_ld;                    // [h][g_ld] iff w_lds
  const bool w_lds = h <= 128;
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int b0 = blockIdx.x * 16;
  const int dir = blockIdx.y;
  const bool reverse = dir ? true : rev0;
  const int gx_off = dir * 4 * h;
  const int hs_off = dir * h;
  const long dbase = (long)dir * B * L;
  const bf16* wsrc = w_hh + (long)dir * 4 * h * h;
  int w_ld = 4 * h;
  if (w_lds) {
    for (int i = threadIdx.x; i < 4 * h * h / 8; i += blockDim.x) {
      const int r = (i * 8) / (4 * h);
      const int c = (i * 8) % (4 * h);
      *reinterpret_cast<s16x8*>(w_s + (long)r * g_ld + c) =
          reinterpret_cast<const s16x8*>(wsrc)[i];
    }
    wsrc = w_s;
    w_ld = g_ld;
  }
  __syncthreads();

  const int N16 = h / 16;
  const int NQ = (N16 + 3) / 4;
  const int lrow = ((lane >> 4) << 2);
  float dc_reg[4][NQ_MAX] = {};
  float dh_reg[4][NQ_MAX] = {};
  int mylen[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int b = b0 + lrow + r;
    mylen[r] = (b < B) ? lens[b] : 0;
  }

  for (int step = L - 1; step >= 0; --step) {
    const int t = reverse ? (L - 1 - step) : step;  // reverse of fwd order
    const int t_prev = reverse ? (t + 1) : (t - 1);
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int b = b0 + lrow + r;
      const bool inb = b < B;
      const bool valid = inb && (t < mylen[r]);
      const long gbase = inb ? ((dbase + (long)b * L + t) * 4 * h) : 0;
      const long xbase = inb ? (((long)b * L + t) * gxs + gx_off) : 0;
#pragma unroll
      for (int q = 0; q < NQ_MAX; ++q) {
        const int f = wid + 4 * q;
        if (q >= NQ || f >= N16) continue;
        const int j = f * 16 + (lane & 15);
        float dgi = 0.f, dgf = 0.f, dgg = 0.f, dgo = 0.f;
        if (valid) {
          const long obase = (dbase + (long)b * L + t) * h + j;
          // interleaved layout: one f32x4 load
          const f32x4 g4 =
              *reinterpret_cast<const f32x4*>(gates + gbase + (long)j * 4);
          const float gi = g4[0], gf = g4[1], gg = g4[2], go = g4[3];
          const float c_t = cs[obase];
          const float c_prev =
              (t_prev >= 0 && t_prev < L && t_prev < mylen[r])
                  ? cs[(dbase + (long)b * L + t_prev) * h + j]
                  : 0.f;
          const float ac = act_f(c_t, relu);
          const float dh =
              dh_reg[r][q] + to_f32(dhs[((long)b * L + t) * hss + hs_off + j]);
          float dc = dc_reg[r][q] + dh * go * dact_from_out(ac, relu);
          dgo = dh * ac * go * (1.f - go);
          // clamp boundary: no grad through a clipped cell state
          if (cell_clip > 0.f && fabsf(c_t) >= cell_clip) dc = 0.f;
          dgi = dc * gg * gi * (1.f - gi);
          dgf = dc * c_prev * gf * (1.f - gf);
          dgg = dc * gi * dact_from_out(gg, relu);
          dc_reg[r][q] = dc * gf;  // carry to previous step
        }
        // write pre-activation gate grads (global + LDS for the MFMA)
        if (inb) {
          from_f32(dgi, &dgates_x[xbase + 0 * h + j]);
          from_f32(dgf, &dgates_x[xbase + 1 * h + j]);
          from_f32(dgg, &dgates_x[xbase + 2 * h + j]);
          from_f32(dgo, &dgates_x[xbase + 3 * h + j]);
        }
        const int lr = lrow + r;
        dg_s[lr * g_ld + 0 * h + j] = __float2bfloat16(dgi);
        dg_s[lr * g_ld + 1 * h + j] = __float2bfloat16(dgf);
        dg_s[lr * g_ld + 2 * h + j] = __float2bfloat16(dgg);
        dg_s[lr * g_ld + 3 * h + j] = __float2bfloat16(dgo);
      }
    }
    __syncthreads();
    // dh_prev = dgates @ W_hh^T : [16,4h] @ [4h,h]; col frag f = wid+4q
    cfrag acc[NQ_MAX];
#pragma unroll
    for (int q = 0; q < NQ_MAX; ++q) acc[q] = cfrag{0.f, 0.f, 0.f, 0.f};
    for (int kk = 0; kk < 4 * h / 32; ++kk) {
      const bfrag adg = lds_frag_l(dg_s, 0, g_ld, kk * 32);
#pragma unroll
      for (int q = 0; q < NQ_MAX; ++q) {
        const int f = wid + 4 * q;
        if (q >= NQ || f >= N16) continue;
        acc[q] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            adg, lds_frag_l(wsrc, f * 16, w_ld, kk * 32), acc[q], 0, 0, 0);
      }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const bool valid = t < mylen[r];
#pragma unroll
      for (int q = 0; q < NQ_MAX; ++q)
        if (valid) dh_reg[r][q] = acc[q][r];
      // invalid step: h passed through unchanged -> dh carries unchanged
    }
    __syncthreads();  // dg_s reads done before next step overwrites
  }
}

// ===================================================================== host
static size_t lstm_fwd_smem(int h, bool w_lds) {
  return ((size_t)16 * (h + 8) + (size_t)16 * 4 * h +
          (w_lds ? (size_t)4 * h * (h + 8) : 0)) * sizeof(bf16);
}
static size_t lstm_bwd_smem(int h, bool w_lds) {
  return ((size_t)16 * (4 * h + 8) + (w_lds ? (size_t)h * (4 * h + 8) : 0))
         * sizeof(bf16);
}

// Bidirectional fused path: gates_x [B,L,8h] (fw|bw), w_hh_t2 [2,4h,h].
// Returns hs [B,L,2h] (concat free), cs [2,B,L,h], gates [2,B,L,h,4].
std::vector<at::Tensor> bilstm_fwd_l(const at::Tensor& gates_x,
                                     const at::Tensor& w_hh_t2,
                                     const at::Tensor& lens, bool relu,
                                     double cell_clip) {
  CHECK_CUDA_CONTIG(gates_x);
  const int B = gates_x.size(0), L = gates_x.size(1);
  const int h = gates_x.size(2) / 8;
  TORCH_CHECK(h % 32 == 0 && h <= H_MAX,
              "bilstm kernel: hidden must be a multiple of 32 and <= 256, got ",
              h);
  TORCH_CHECK(w_hh_t2.size(0) == 2 && w_hh_t2.scalar_type() == at::kBFloat16,
              "w_hh_t2 must be bf16 [2,4h,h]");
  auto hs = at::empty({B, L, 2 * h}, gates_x.options());
  auto cs = at::empty({2, B, L, h}, gates_x.options().dtype(at::kFloat));
  auto gates = at::empty({2, B, L, 4 * h},
                         gates_x.options().dtype(at::kFloat));
  const size_t smem = lstm_fwd_smem(h, h <= 128);
  TORCH_CHECK(smem <= 160 * 1024, "lstm fwd LDS overflow");
  const dim3 grid((B + 15) / 16, 2);
  auto stream = cur_stream(gates_x);
  if (gates_x.scalar_type() == at::kBFloat16) {
    hipLaunchKernelGGL(lstm_fwd_kernel<bf16>, grid, dim3(256), smem, stream,
                       (const bf16*)gates_x.data_ptr(),
                       (const bf16*)w_hh_t2.data_ptr(), lens.data_ptr<int>(),
                       (bf16*)hs.data_ptr(), cs.data_ptr<float>(),
                       gates.data_ptr<float>(), B, L, h, 8 * h, 2 * h,
                       /*rev0=*/false, relu, (float)cell_clip);
  } else {
    hipLaunchKernelGGL(lstm_fwd_kernel<float>, grid, dim3(256), smem, stream,
                       gates_x.data_ptr<float>(),
                       (const bf16*)w_hh_t2.data_ptr(), lens.data_ptr<int>(),
                       hs.data_ptr<float>(), cs.data_ptr<float>(),
                       gates.data_ptr<float>(), B, L, h, 8 * h, 2 * h,
                       /*rev0=*/false, relu, (float)cell_clip);
  }
  HIP_CHECK_LAST();
  return {hs, cs, gates};
}

at::Tensor bilstm_bwd_l(const at::Tensor& dhs, const at::Tensor& cs,
                        const at::Tensor& gates, const at::Tensor& w_hh2,
                        const at::Tensor& lens, bool relu, double cell_clip) {
  const int B = dhs.size(0), L = dhs.size(1);
  const int h = dhs.size(2) / 2;
  TORCH_CHECK(w_hh2.size(0) == 2 && w_hh2.scalar_type() == at::kBFloat16,
              "w_hh2 must be bf16 [2,h,4h]");
  auto dhs_c = dhs.contiguous();
  auto dgates_x = at::empty({B, L, 8 * h}, dhs.options());
  const size_t smem = lstm_bwd_smem(h, h <= 128);
  TORCH_CHECK(smem <= 160 * 1024, "lstm bwd LDS overflow");
  const dim3 grid((B + 15) / 16, 2);
  auto stream = cur_stream(dhs);
  if (dhs.scalar_type() == at::kBFloat16) {
    hipLaunchKernelGGL(lstm_bwd_kernel<bf16>, grid, dim3(256), smem, stream,
                       (const bf16*)dhs_c.data_ptr(), cs.data_ptr<float>(),
                       gates.data_ptr<float>(), (const bf16*)w_hh2.data_ptr(),
                       lens.data_ptr<int>(), (bf16*)dgates_x.data_ptr(), B, L,
                       h, 8 * h, 2 * h, /*rev0=*/false, relu, (float)cell_clip);
  } else {
    hipLaunchKernelGGL(lstm_bwd_kernel<float>, grid, dim3(256), smem, stream,
                       dhs_c.data_ptr<float>(), cs.data_ptr<float>(),
                       gates.data_ptr<float>(), (const bf16*)w_hh2.data_ptr(),
                       lens.data_ptr<int>(), dgates_x.data_ptr<float>(), B, L,
                       h, 8 * h, 2 * h, /*rev0=*/false, relu, (float)cell_clip);
  }
  HIP_CHECK_LAST();
  return dgates_x;
}

// Single-direction path (kept for the kernel unit tests: one direction,
// reverse selectable, gates_x [B,L,4h] contiguous).
std::vector<at::Tensor> lstm_fwd(const at::Tensor& gates_x,
                                 const at::Tensor& w_hh,
                                 const at::Tensor& lens, bool reverse,
                                 bool relu) {
  CHECK_CUDA_CONTIG(gates_x);
  const int B = gates_x.size(0), L = gates_x.size(1);
  const int h4 = gates_x.size(2), h = h4 / 4;
  TORCH_CHECK(h % 32 == 0 && h <= H_MAX, "lstm kernel: bad hidden ", h);
  auto hs = at::empty({B, L, h}, gates_x.options());
  auto cs = at::empty({1, B, L, h}, gates_x.options().dtype(at::kFloat));
  auto gates = at::empty({1, B, L, h4}, gates_x.options().dtype(at::kFloat));
  auto w_t = w_hh.t().contiguous().to(at::kBFloat16);  // [4h,h]
  const size_t smem = lstm_fwd_smem(h, h <= 128);
  TORCH_CHECK(smem <= 160 * 1024, "lstm fwd LDS overflow");
  const dim3 grid((B + 15) / 16, 1);
  auto stream = cur_stream(gates_x);
  if (gates_x.scalar_type() == at::kBFloat16) {
    hipLaunchKernelGGL(lstm_fwd_kernel<bf16>, grid, dim3(256), smem, stream,
                       (const bf16*)gates_x.data_ptr(),
                       (const bf16*)w_t.data_ptr(), lens.data_ptr<int>(),
                       (bf16*)hs.data_ptr(), cs.data_ptr<float>(),
                       gates.data_ptr<float>(), B, L, h, h4, h, reverse, relu,
                       0.f);
  } else {
    hipLaunchKernelGGL(lstm_fwd_kernel<float>, grid, dim3(256), smem, stream,
                       gates_x.data_ptr<float>(), (const bf16*)w_t.data_ptr(),
                       lens.data_ptr<int>(), hs.data_ptr<float>(),
                       cs.data_ptr<float>(), gates.data_ptr<float>(), B, L, h,
                       h4, h, reverse, relu, 0.f);
  }
  HIP_CHECK_LAST();
  return {hs, cs.squeeze(0), gates.squeeze(0)};
}

std::vector<at::Tensor> lstm_bwd(const at::Tensor& dhs, const at::Tensor& hs,
                                 const at::Tensor& cs, const at::Tensor& gates,
                                 const at::Tensor& w_hh, const at::Tensor& lens,
                                 bool reverse, bool relu) {
  const int B = dhs.size(0), L = dhs.size(1), h = dhs.size(2);
  auto dgates_x = at::empty({B, L, 4 * h}, dhs.options());
  auto w_b = w_hh.contiguous().to(at::kBFloat16);  // [h,4h]
  const size_t smem = lstm_bwd_smem(h, h <= 128);
  TORCH_CHECK(smem <= 160 * 1024, "lstm bwd LDS overflow");
  const dim3 grid((B + 15) / 16, 1);
  auto stream = cur_stream(dhs);
  auto dhs_c = dhs.contiguous();
  if (dhs.scalar_type() == at::kBFloat16) {
    hipLaunchKernelGGL(lstm_bwd_kernel<bf16>, grid, dim3(256), smem, stream,
                       (const bf16*)dhs_c.data_ptr(), cs.data_ptr<float>(),
                       gates.data_ptr<float>(), (const bf16*)w_b.data_ptr(),
                       lens.data_ptr<int>(), (bf16*)dgates_x.data_ptr(), B, L,
                       h, 4 * h, h, reverse, relu, 0.f);
  } else {
    hipLaunchKernelGGL(lstm_bwd_kernel<float>, grid, dim3(256), smem, stream,
                       dhs_c.data_ptr<float>(), cs.data_ptr<float>(),
                       gates.data_ptr<float>(), (const bf16*)w_b.data_ptr(),
                       lens.data_ptr<int>(), dgates_x.data_ptr<float>(), B, L,
                       h, 4 * h, h, reverse, relu, 0.f);
  }
  HIP_CHECK_LAST();
  return {dgates_x};
}
