// Fused attention fwd/bwd for BERT + vanilla MHA and
// the TENER relative-position variant (K9), hand-written for gfx950:
// MFMA 16x16x32 bf16, all tiles LDS-resident (reference seq lens are
// <= 170, so K/V/P fit in the 160 KiB LDS whole -- no online softmax
// needed; kernels assert L_pad <= 176).
//
// Layout notes (guide cdna_hip_programming.md §3):
//  * A-frag  lane l -> A[(l&15)][kk*32 + (l>>4)*8 + e], e = 0..7
//  * B-frag  lane l -> B[kk*32 + (l>>4)*8 + e][(l&15)] — read from an
//    [N][K]-stored (transposed) buffer with the same indexing as A.
//  * C/D     lane l, reg r -> D[(l>>4)*4 + r][nf*16 + (l&15)]
// Key (K) and V^T are staged so every MFMA operand read is one 16-byte
// contiguous LDS read.
#include "common.h"

#define MAXNF 12  // N-tiles of 16: L_pad <= 176 -> 11
#define LPAD_MAX 176

using bfrag = mfma_bf16x8;
using cfrag = mfma_f32x4;

__device__ __forceinline__ bfrag lds_frag(const bf16* base, int i0, int ld,
                                          int k0) {
  const int l = threadIdx.x & (WAVE - 1);
  return *reinterpret_cast<const bfrag*>(base + (long)(i0 + (l & 15)) * ld +
                                         k0 + ((l >> 4) << 3));
}

__device__ __forceinline__ cfrag mfma16(bfrag a, bfrag b, cfrag c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

// stage a [rows x cols] bf16 global tile into LDS row-major, zero-padding
// to [rpad x cols]; cols % 8 == 0
__device__ __forceinline__ void stage_tile(const bf16* g, bf16* s, int rows,
                                           int rpad, int cols) {
  const int nv = rpad * cols / 8;
  for (int i = threadIdx.x; i < nv; i += blockDim.x) {
    const int r = (i * 8) / cols;
    s16x8 val{};
    if (r < rows) val = reinterpret_cast<const s16x8*>(g)[i];
    reinterpret_cast<s16x8*>(s)[i] = val;
  }
}

// stage transposed: global [rows x cols] -> LDS [cols x rpad]
__device__ __forceinline__ void stage_tile_T(const bf16* g, bf16* s, int rows,
                                             int rpad, int cols) {
  for (int i = threadIdx.x; i < rpad * cols / 8; i += blockDim.x) {
    const int r = (i * 8) / cols;
    const int c0 = (i * 8) % cols;
    s16x8 val{};
    if (r < rows) val = reinterpret_cast<const s16x8*>(g)[i];
#pragma unroll
    for (int e = 0; e < 8; ++e)
      s[(long)(c0 + e) * rpad + r] = reinterpret_cast<const bf16*>(&val)[e];
  }
}

// ---------------------------------------------------------------------
// forward: one block = one (b,h); 4 waves each loop M-tiles of 16 rows.
// out[b,h] = softmax(scale * Q K^T + keymask) V ; lse saved for bwd.
// ---------------------------------------------------------------------
__global__ __launch_bounds__(256) void attn_fwd_kernel(
    const bf16* __restrict__ q, const bf16* __restrict__ k,
    const bf16* __restrict__ v, const int* __restrict__ lens,
    bf16* __restrict__ out, float* __restrict__ lse, int B, int H, int L,
    int D, int Lpad, float scale) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  bf16* q_s = reinterpret_cast<bf16*>(smem_raw);    // [Lpad][D]
  bf16* k_s = q_s + Lpad * D;                       // [Lpad][D]
  bf16* vt_s = k_s + Lpad * D;                      // [D][Lpad]
  bf16* p_s = vt_s + D * Lpad;                      // [4 waves][16][Lpad]

  const int bh = blockIdx.x;
  const int b = bh / H;
  const long base = (long)bh * L * D;
  const int len = lens[b];
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);

  stage_tile(q + base, q_s, L, Lpad, D);
  stage_tile(k + base, k_s, L, Lpad, D);
  stage_tile_T(v + base, vt_s, L, Lpad, D);
  __syncthreads();

  const int NF = Lpad / 16;
  const int NKK = D / 32;
  bf16* pw = p_s + wid * 16 * Lpad;

  for (int m0 = wid * 16; m0 < L; m0 += 4 * 16) {
    // ---- S = scale * Q K^T over this wave's 16 rows ----
    cfrag acc[MAXNF];
#pragma unroll
    for (int nf = 0; nf < MAXNF; ++nf) acc[nf] = cfrag{0.f, 0.f, 0.f, 0.f};
    bfrag aq[2];
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
      if (kk < NKK) aq[kk] = lds_frag(q_s, m0, D, kk * 32);
#pragma unroll
    for (int nf = 0; nf < MAXNF; ++nf) {
      if (nf >= NF) break;
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
        if (kk < NKK)
          acc[nf] = mfma16(aq[kk], lds_frag(k_s, nf * 16, D, kk * 32), acc[nf]);
    }
    // ---- softmax rows (C layout: reg r = row, lanes&15 = col) ----
    float row_lse[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float mx = -1e30f;
#pragma unroll
      for (int nf = 0; nf < MAXNF; ++nf) {
        if (nf >= NF) break;
        const int col = nf * 16 + (lane & 15);
        const float s = (col < len) ? acc[nf][r] * scale : -1e30f;
        acc[nf][r] = s;
        mx = fmaxf(mx, s);
      }
      mx = group16_reduce_max(mx);
      float sum = 0.f;
#pragma unroll
      for (int nf = 0; nf < MAXNF; ++nf) {
        if (nf >= NF) break;
        const float p = __expf(acc[nf][r] - mx);
        acc[nf][r] = p;
        sum += p;
      }
      sum = group16_reduce_sum(sum);
      const float inv = __frcp_rn(sum);
#pragma unroll
      for (int nf = 0; nf < MAXNF; ++nf) {
        if (nf >= NF) break;
        acc[nf][r] *= inv;
      }
      row_lse[r] = mx + __logf(sum);
    }
    // write normalized P (bf16) into this wave's LDS tile [16][Lpad]
#pragma unroll
    for (int nf = 0; nf < MAXNF; ++nf) {
      if (nf >= NF) break;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int lr = ((lane >> 4) << 2) + r;
        pw[lr * Lpad + nf * 16 + (lane & 15)] = __float2bfloat16(acc[nf][r]);
      }
    }
    if ((lane & 15) == 0) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = m0 + ((lane >> 4) << 2) + r;
        if (row < L) lse[(long)bh * L + row] = row_lse[r];
      }
    }
    // ---- O = P V  (A from pw, B from vt_s) ----
    const int NFD = D / 16;
    cfrag oacc[4];
#pragma unroll
    for (int i = 0; i < 4; ++i) oacc[i] = cfrag{0.f, 0.f, 0.f, 0.f};
    for (int kk = 0; kk < Lpad / 32; ++kk) {
      const bfrag ap = lds_frag(pw, 0, Lpad, kk * 32);
#pragma unroll
      for (int nd = 0; nd < 4; ++nd)
        if (nd < NFD)
          oacc[nd] = mfma16(ap, lds_frag(vt_s, nd * 16, Lpad, kk * 32), oacc[nd]);
    }
#pragma unroll
    for (int nd = 0; nd < 4; ++nd) {
      if (nd >= NFD) break;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = m0 + ((lane >> 4) << 2) + r;
        if (row < L)
          out[base + (long)row * D + nd * 16 + (lane & 15)] =
              __float2bfloat16(oacc[nd][r]);
      }
    }
  }
}

// ---------------------------------------------------------------------
// backward: one block = one (b,h).
// P recomputed from lse; dV = P^T dO ; dS = P*(dP - delta); dQ = dS K ;
// dK = dS^T Q. P^T lives in LDS and is overwritten by dS^T in place.
// ---------------------------------------------------------------------
__global__ __launch_bounds__(256) void attn_bwd_kernel(
    const bf16* __restrict__ dout, const bf16* __restrict__ q,
    const bf16* __restrict__ k, const bf16* __restrict__ v,
    const bf16* __restrict__ o, const float* __restrict__ lse,
    const int* __restrict__ lens, bf16* __restrict__ dq,
    bf16* __restrict__ dk, bf16* __restrict__ dv, int B, int H, int L, int D,
    int Lpad, float scale) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  bf16* q_s = reinterpret_cast<bf16*>(smem_raw);   // [Lpad][D]
  bf16* k_s = q_s + Lpad * D;                      // [Lpad][D]
  bf16* v_s = k_s + Lpad * D;                      // [Lpad][D] (row major!)
  bf16* do_s = v_s + Lpad * D;                     // [Lpad][D]
  bf16* pt_s = do_s + Lpad * D;                    // [Lpad][Lpad]  P^T / dS^T
  float* delta_s = reinterpret_cast<float*>(pt_s + (long)Lpad * Lpad);  // [Lpad]
  float* lse_s = delta_s + Lpad;                                        // [Lpad]

  const int bh = blockIdx.x;
  const int b = bh / H;
  const long base = (long)bh * L * D;
  const int len = lens[b];
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int NF = Lpad / 16;
  const int NKK = D / 32;
  const int NFD = D / 16;

  stage_tile(q + base, q_s, L, Lpad, D);
  stage_tile(k + base, k_s, L, Lpad, D);
  stage_tile(v + base, v_s, L, Lpad, D);
  stage_tile(dout + base, do_s, L, Lpad, D);
  // delta[row] = sum_d dO[row,d] * O[row,d]; lse -> LDS
  for (int r = threadIdx.x; r < Lpad; r += blockDim.x) {
    float acc = 0.f;
    if (r < L) {
      for (int d = 0; d < D; ++d)
        acc += to_f32(dout[base + (long)r * D + d]) *
               to_f32(o[base + (long)r * D + d]);
      lse_s[r] = lse[(long)bh * L + r];
    } else {
      lse_s[r] = 0.f;
    }
    delta_s[r] = acc;
  }
  __syncthreads();

  // ---- phase 1: recompute P, store P^T ----
  for (int m0 = wid * 16; m0 < Lpad; m0 += 64) {
    cfrag acc[MAXNF];
#pragma unroll
    for (int nf = 0; nf < MAXNF; ++nf) acc[nf] = cfrag{0.f, 0.f, 0.f, 0.f};
    bfrag aq[2];
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
      if (kk < NKK) aq[kk] = lds_frag(q_s, m0, D, kk * 32);
#pragma unroll
    for (int nf = 0; nf < MAXNF; ++nf) {
      if (nf >= NF) break;
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
        if (kk < NKK)
          acc[nf] = mfma16(aq[kk], lds_frag(k_s, nf * 16, D, kk * 32), acc[nf]);
    }
#pragma unroll
    for (int nf = 0; nf < MAXNF; ++nf) {
      if (nf >= NF) break;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = m0 + ((lane >> 4) << 2) + r;
        const int col = nf * 16 + (lane & 15);
        float p = 0.f;
        if (row < L && col < len)
          p = __expf(acc[nf][r] * scale - lse_s[row]);
        pt_s[(long)col * Lpad + row] = __float2bfloat16(p);
      }
    }
  }
  __syncthreads();

  // ---- phase 2: dV[kt] = P^T[kt,:] dO  (key tiles round-robin) ----
  for (int k0 = wid * 16; k0 < L; k0 += 64) {
    cfrag acc[4];
#pragma unroll
    for (int i = 0; i < 4; ++i) acc[i] = cfrag{0.f, 0.f, 0.f, 0.f};
    for (int kk = 0; kk < Lpad / 32; ++kk) {
      const bfrag ap = lds_frag(pt_s, k0, Lpad, kk * 32);
#pragma unroll
      for (int nd = 0; nd < 4; ++nd)
        if (nd < NFD) {
          // B = dO[rows, D] : B[k=row][col=d] contiguous in do_s? No:
          // do_s is [row][d] row-major, element e varies row -> strided.
          // Use transposed read via per-element loads.
          bfrag bo;
#pragma unroll
          for (int e = 0; e < 8; ++e)
            reinterpret_cast<bf16*>(&bo)[e] =
                do_s[(long)(kk * 32 + ((lane >> 4) << 3) + e) * D + nd * 16 +
                     (lane & 15)];
          acc[nd] = mfma16(ap, bo, acc[nd]);
        }
    }
#pragma unroll
    for (int nd = 0; nd < 4; ++nd) {
      if (nd >= NFD) break;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int krow = k0 + ((lane >> 4) << 2) + r;
        if (krow < L)
          dv[base + (long)krow * D + nd * 16 + (lane & 15)] =
              __float2bfloat16(acc[nd][r]);
      }
    }
  }
  __syncthreads();

  // ---- phase 3: dP = dO V^T ; dS = P (dP - delta) * scale -> dS^T ----
  for (int m0 = wid * 16; m0 < L; m0 += 64) {
    cfrag acc[MAXNF];
#pragma unroll
    for (int nf = 0; nf < MAXNF; ++nf) acc[nf] = cfrag{0.f, 0.f, 0.f, 0.f};
    bfrag ado[2];
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
      if (kk < NKK) ado[kk] = lds_frag(do_s, m0, D, kk * 32);
#pragma unroll
    for (int nf = 0; nf < MAXNF; ++nf) {
      if (nf >= NF) break;
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
        if (kk < NKK)
          acc[nf] = mfma16(ado[kk], lds_frag(v_s, nf * 16, D, kk * 32), acc[nf]);
    }
#pragma unroll
    for (int nf = 0; nf < MAXNF; ++nf) {
      if (nf >= NF) break;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = m0 + ((lane >> 4) << 2) + r;
        const int col = nf * 16 + (lane & 15);
        const float p = to_f32(pt_s[(long)col * Lpad + row]);
        const float ds = p * (acc[nf][r] - delta_s[row]) * scale;
        // do NOT write yet — in-place overwrite of P^T is safe only
        // element-wise within this lane's own slots, which it is:
        pt_s[(long)col * Lpad + row] = __float2bfloat16(ds);
      }
    }
  }
  __syncthreads();

  // ---- phase 4: dK[kt] = dS^T[kt,:] Q ----
  for (int k0 = wid * 16; k0 < L; k0 += 64) {
    cfrag acc[4];
#pragma unroll
    for (int i = 0; i < 4; ++i) acc[i] = cfrag{0.f, 0.f, 0.f, 0.f};
    for (int kk = 0; kk < Lpad / 32; ++kk) {
      const bfrag ads = lds_frag(pt_s, k0, Lpad, kk * 32);
#pragma unroll
      for (int nd = 0; nd < 4; ++nd)
        if (nd < NFD) {
          bfrag bq;
#pragma unroll
          for (int e = 0; e < 8; ++e)
            reinterpret_cast<bf16*>(&bq)[e] =
                q_s[(long)(kk * 32 + ((lane >> 4) << 3) + e) * D + nd * 16 +
                    (lane & 15)];
          acc[nd] = mfma16(ads, bq, acc[nd]);
        }
    }
#pragma unroll
    for (int nd = 0; nd < 4; ++nd) {
      if (nd >= NFD) break;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int krow = k0 + ((lane >> 4) << 2) + r;
        if (krow < L)
          dk[base + (long)krow * D + nd * 16 + (lane & 15)] =
              __float2bfloat16(acc[nd][r]);
      }
    }
  }

  // ---- phase 5: dQ = dS K  (A = dS rows — strided reads from dS^T) ----
  for (int m0 = wid * 16; m0 < L; m0 += 64) {
    cfrag acc[4];
#pragma unroll
    for (int i = 0; i < 4; ++i) acc[i] = cfrag{0.f, 0.f, 0.f, 0.f};
    for (int kk = 0; kk < Lpad / 32; ++kk) {
      bfrag ads;
#pragma unroll
      for (int e = 0; e < 8; ++e)
        reinterpret_cast<bf16*>(&ads)[e] =
            pt_s[(long)(kk * 32 + ((lane >> 4) << 3) + e) * Lpad + m0 +
                 (lane & 15)];
#pragma unroll
      for (int nd = 0; nd < 4; ++nd)
        if (nd < NFD) {
          // B = K[key, d]: row-major [key][d] needs per-element reads
          // (the lds_frag helper would read it as K^T)
          bfrag bk;
#pragma unroll
          for (int e = 0; e < 8; ++e)
            reinterpret_cast<bf16*>(&bk)[e] =
                k_s[(long)(kk * 32 + ((lane >> 4) << 3) + e) * D + nd * 16 +
                    (lane & 15)];
          acc[nd] = mfma16(ads, bk, acc[nd]);
        }
    }
#pragma unroll
    for (int nd = 0; nd < 4; ++nd) {
      if (nd >= NFD) break;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = m0 + ((lane >> 4) << 2) + r;
        if (row < L)
          dq[base + (long)row * D + nd * 16 + (lane & 15)] =
              __float2bfloat16(acc[nd][r]);
      }
    }
  }
}

// ===================================================================== host
static void check_attn_args(const at::Tensor& q, int& B, int& H, int& L,
                            int& D, int& Lpad) {
  CHECK_CUDA_CONTIG(q);
  TORCH_CHECK(q.scalar_type() == at::kBFloat16, "attention: bf16 only");
  B = q.size(0);
  H = q.size(1);
  L = q.size(2);
  D = q.size(3);
  TORCH_CHECK(D == 32 || D == 64, "attention: head dim must be 32 or 64 "
              "(pad in the wrapper), got ", D);
  Lpad = ((L + 31) / 32) * 32;
  TORCH_CHECK(Lpad <= LPAD_MAX,
              "attention: seq len > 176 unsupported by the full-LDS kernel");
}

std::vector<at::Tensor> attn_fwd(const at::Tensor& q, const at::Tensor& k,
                                 const at::Tensor& v, const at::Tensor& lens,
                                 double scale) {
  int B, H, L, D, Lpad;
  check_attn_args(q, B, H, L, D, Lpad);
  auto out = at::empty_like(q);
  auto lse = at::empty({B, H, L}, q.options().dtype(at::kFloat));
  const size_t smem = (size_t)(3 * Lpad * D + 4 * 16 * Lpad) * sizeof(bf16);
  TORCH_CHECK(smem <= 160 * 1024, "attn fwd LDS overflow");
  hipLaunchKernelGGL(attn_fwd_kernel, dim3(B * H), dim3(256), smem,
                     cur_stream(q), (const bf16*)q.data_ptr(),
                     (const bf16*)k.data_ptr(), (const bf16*)v.data_ptr(),
                     lens.data_ptr<int>(), (bf16*)out.data_ptr(),
                     lse.data_ptr<float>(), B, H, L, D, Lpad, (float)scale);
  HIP_CHECK_LAST();
  return {out, lse};
}

std::vector<at::Tensor> attn_bwd(const at::Tensor& dout, const at::Tensor& q,
                                 const at::Tensor& k, const at::Tensor& v,
                                 const at::Tensor& o, const at::Tensor& lse,
                                 const at::Tensor& lens, double scale) {
  int B, H, L, D, Lpad;
  check_attn_args(q, B, H, L, D, Lpad);
  auto dq = at::empty_like(q);
  auto dk = at::empty_like(k);
  auto dv = at::empty_like(v);
  const size_t smem = (size_t)(4 * Lpad * D + (long)Lpad * Lpad) * sizeof(bf16)
                      + 2 * Lpad * sizeof(float);
  TORCH_CHECK(smem <= 160 * 1024, "attn bwd LDS overflow");
  hipLaunchKernelGGL(attn_bwd_kernel, dim3(B * H), dim3(256), smem,
                     cur_stream(q), (const bf16*)dout.data_ptr(),
                     (const bf16*)q.data_ptr(), (const bf16*)k.data_ptr(),
                     (const bf16*)v.data_ptr(), (const bf16*)o.data_ptr(),
                     lse.data_ptr<float>(), lens.data_ptr<int>(),
                     (bf16*)dq.data_ptr(), (bf16*)dk.data_ptr(),
                     (bf16*)dv.data_ptr(), B, H, L, D, Lpad, (float)scale);
  HIP_CHECK_LAST();
  return {dq, dk, dv};
}
