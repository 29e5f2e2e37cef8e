// TENER relative-position attention (the rel-pos kernel
// named in BASELINE.json): S[i,j] = (q_i+u)·k_j + (q_i+v)·R[j-i+L-1],
// unscaled, no key projection (reference tools/transformer/tener.py:12-74).
// The reference's zero-pad "shift" trick is replaced by computing the
// full BD = (q+v) R^T tile with MFMA and gathering BD[i, j-i+L-1]
// directly from LDS (direct indexing).
//
// The wrapper passes qu = q+u and qv = q+v (u/v grads are reductions of
// dqu/dqv done in python) with head dim zero-padded to 32. L_pad <= 160.
#include "common.h"

#define TMAXNF 10   // Lpad/16 <= 10
#define TLPAD_MAX 160

using bfrag = mfma_bf16x8;
using cfrag = mfma_f32x4;

__device__ __forceinline__ bfrag lds_frag_t(const bf16* base, int i0, int ld,
                                            int k0) {
  const int l = threadIdx.x & (WAVE - 1);
  return *reinterpret_cast<const bfrag*>(base + (long)(i0 + (l & 15)) * ld +
                                         k0 + ((l >> 4) << 3));
}

__device__ __forceinline__ cfrag mfma16t(bfrag a, bfrag b, cfrag c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

__device__ __forceinline__ void stage_t(const bf16* g, bf16* s, int rows,
                                        int rpad, int cols) {
  for (int i = threadIdx.x; i < rpad * cols / 8; i += blockDim.x) {
    const int r = (i * 8) / cols;
    s16x8 val{};
    if (r < rows) val = reinterpret_cast<const s16x8*>(g)[i];
    reinterpret_cast<s16x8*>(s)[i] = val;
  }
}

__device__ __forceinline__ void stage_t_T(const bf16* g, bf16* s, int rows,
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

// one block per (b,h); D = 32 (padded head dim)
__global__ __launch_bounds__(256) void tener_fwd_kernel(
    const bf16* __restrict__ qu, const bf16* __restrict__ qv,
    const bf16* __restrict__ k, const bf16* __restrict__ v,
    const bf16* __restrict__ rel,  // [2L, D]
    const int* __restrict__ lens, bf16* __restrict__ out,
    float* __restrict__ lse, int B, int H, int L, int D, int Lpad) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  bf16* qu_s = reinterpret_cast<bf16*>(smem_raw);   // [Lpad][D]
  bf16* k_s = qu_s + Lpad * D;                      // [Lpad][D]
  bf16* vt_s = k_s + Lpad * D;                      // [D][Lpad]
  bf16* r_s = vt_s + D * Lpad;                      // [2Lpad][D]
  bf16* bd_s = r_s + 2 * Lpad * D;                  // [4][16][2Lpad]
  bf16* p_s = bd_s + 4 * 16 * 2 * Lpad;             // [4][16][Lpad]

  const int bh = blockIdx.x;
  const int b = bh / H;
  const long base = (long)bh * L * D;
  const int len = lens[b];
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);

  stage_t(qu + base, qu_s, L, Lpad, D);
  stage_t(k + base, k_s, L, Lpad, D);
  stage_t_T(v + base, vt_s, L, Lpad, D);
  stage_t(rel, r_s, 2 * L, 2 * Lpad, D);
  __syncthreads();

  const int NF = Lpad / 16;
  bf16* pw = p_s + wid * 16 * Lpad;
  bf16* bdw = bd_s + wid * 16 * 2 * Lpad;

  for (int m0 = wid * 16; m0 < L; m0 += 64) {
    // ---- BD tile: (q+v) R^T -> LDS [16][2Lpad] (qv A-frag from global)
    bfrag aqv;
    {
      const int l = lane;
      aqv = *reinterpret_cast<const bfrag*>(
          qv + base + (long)(min(m0 + (l & 15), L - 1)) * D + ((l >> 4) << 3));
      if (m0 + (lane & 15) >= L) aqv = bfrag{};
    }
    for (int nf2 = 0; nf2 < 2 * NF; ++nf2) {
      cfrag acc = cfrag{0.f, 0.f, 0.f, 0.f};
      acc = mfma16t(aqv, lds_frag_t(r_s, nf2 * 16, D, 0), acc);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int lr = ((lane >> 4) << 2) + r;
        bdw[(long)lr * 2 * Lpad + nf2 * 16 + (lane & 15)] =
            __float2bfloat16(acc[r]);
      }
    }
    // ---- S = qu k^T + gather(BD) ----
    cfrag acc[TMAXNF];
#pragma unroll
    for (int nf = 0; nf < TMAXNF; ++nf) acc[nf] = cfrag{0.f, 0.f, 0.f, 0.f};
    const bfrag aq = lds_frag_t(qu_s, m0, D, 0);
#pragma unroll
    for (int nf = 0; nf < TMAXNF; ++nf) {
      if (nf < NF) {
      acc[nf] = mfma16t(aq, lds_frag_t(k_s, nf * 16, D, 0), acc[nf]);
      }
    }
    float row_lse[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int lr = ((lane >> 4) << 2) + r;
      const int row = m0 + lr;
      float mx = -1e30f;
#pragma unroll
      for (int nf = 0; nf < TMAXNF; ++nf) {
        if (nf < NF) {
        const int col = nf * 16 + (lane & 15);
        float s = -1e30f;
        if (col < len && row < L) {
          const int off = col - row + L - 1;  // in [0, 2L)
          s = acc[nf][r] + to_f32(bdw[(long)lr * 2 * Lpad + off]);
        }
        acc[nf][r] = s;
        mx = fmaxf(mx, s);
        }
      }
      mx = group16_reduce_max(mx);
      float sum = 0.f;
#pragma unroll
      for (int nf = 0; nf < TMAXNF; ++nf) {
        if (nf < NF) {
        const float p = __expf(acc[nf][r] - mx);
        acc[nf][r] = p;
        sum += p;
        }
      }
      sum = group16_reduce_sum(sum);
      const float inv = __frcp_rn(sum);
#pragma unroll
      for (int nf = 0; nf < TMAXNF; ++nf) {
        if (nf < NF) {
        acc[nf][r] *= inv;
        }
      }
      row_lse[r] = mx + __logf(sum);
    }
#pragma unroll
    for (int nf = 0; nf < TMAXNF; ++nf) {
      if (nf < NF) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int lr = ((lane >> 4) << 2) + r;
        pw[lr * Lpad + nf * 16 + (lane & 15)] = __float2bfloat16(acc[nf][r]);
      }
      }
    }
    if ((lane & 15) == 0) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = m0 + ((lane >> 4) << 2) + r;
        if (row < L) lse[(long)bh * L + row] = row_lse[r];
      }
    }
    // ---- O = P V ----
    cfrag oacc[2];
#pragma unroll
    for (int i = 0; i < 2; ++i) oacc[i] = cfrag{0.f, 0.f, 0.f, 0.f};
    for (int kk = 0; kk < Lpad / 32; ++kk) {
      const bfrag ap = lds_frag_t(pw, 0, Lpad, kk * 32);
#pragma unroll
      for (int nd = 0; nd < 2; ++nd)
        oacc[nd] = mfma16t(ap, lds_frag_t(vt_s, nd * 16, Lpad, kk * 32),
                           oacc[nd]);
    }
#pragma unroll
    for (int nd = 0; nd < 2; ++nd)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = m0 + ((lane >> 4) << 2) + r;
        if (row < L)
          out[base + (long)row * D + nd * 16 + (lane & 15)] =
              __float2bfloat16(oacc[nd][r]);
      }
  }
}

// backward: dV = P^T dO; dS = P(dP - delta); dK = dS^T qu;
// dqu = dS k; dqv = dBD R with dBD[i,m] = dS[i, m-(L-1)+i].
__global__ __launch_bounds__(256) void tener_bwd_kernel(
    const bf16* __restrict__ dout, const bf16* __restrict__ qu,
    const bf16* __restrict__ qv, const bf16* __restrict__ k,
    const bf16* __restrict__ v, const bf16* __restrict__ rel,
    const bf16* __restrict__ o, const float* __restrict__ lse,
    const int* __restrict__ lens, bf16* __restrict__ dqu,
    bf16* __restrict__ dqv, bf16* __restrict__ dk, bf16* __restrict__ dv,
    int B, int H, int L, int D, int Lpad) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  bf16* qu_s = reinterpret_cast<bf16*>(smem_raw);  // [Lpad][D]
  bf16* k_s = qu_s + Lpad * D;                     // [Lpad][D]
  bf16* do_s = k_s + Lpad * D;                     // [Lpad][D]
  bf16* r_s = do_s + Lpad * D;                     // [2Lpad][D]
  // union region: phase A stages v row-major + per-wave BD tiles; phase E
  // reuses it as per-wave dBD tiles
  bf16* un_s = r_s + 2 * Lpad * D;                 // [4][16][2Lpad]
  bf16* v_s = un_s + 4 * 16 * 2 * Lpad;            // [Lpad][D]
  bf16* pt_s = v_s + Lpad * D;                     // [Lpad][Lpad]
  float* delta_s = reinterpret_cast<float*>(pt_s + (long)Lpad * Lpad);
  float* lse_s = delta_s + Lpad;

  const int bh = blockIdx.x;
  const int b = bh / H;
  const long base = (long)bh * L * D;
  const int len = lens[b];
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int NF = Lpad / 16;

  stage_t(qu + base, qu_s, L, Lpad, D);
  stage_t(k + base, k_s, L, Lpad, D);
  stage_t(dout + base, do_s, L, Lpad, D);
  stage_t(rel, r_s, 2 * L, 2 * Lpad, D);
  stage_t(v + base, v_s, L, Lpad, D);
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

  bf16* bdw = un_s + wid * 16 * 2 * Lpad;

  // ---- phase 1: recompute P -> P^T ----
  for (int m0 = wid * 16; m0 < Lpad; m0 += 64) {
    bfrag aqv;
    {
      const int l = lane;
      aqv = *reinterpret_cast<const bfrag*>(
          qv + base + (long)(min(m0 + (l & 15), L - 1)) * D + ((l >> 4) << 3));
      if (m0 + (lane & 15) >= L) aqv = bfrag{};
    }
    for (int nf2 = 0; nf2 < 2 * NF; ++nf2) {
      cfrag acc = cfrag{0.f, 0.f, 0.f, 0.f};
      acc = mfma16t(aqv, lds_frag_t(r_s, nf2 * 16, D, 0), acc);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int lr = ((lane >> 4) << 2) + r;
        bdw[(long)lr * 2 * Lpad + nf2 * 16 + (lane & 15)] =
            __float2bfloat16(acc[r]);
      }
    }
    cfrag acc[TMAXNF];
#pragma unroll
    for (int nf = 0; nf < TMAXNF; ++nf) acc[nf] = cfrag{0.f, 0.f, 0.f, 0.f};
    const bfrag aq = lds_frag_t(qu_s, m0, D, 0);
#pragma unroll
    for (int nf = 0; nf < TMAXNF; ++nf) {
      if (nf < NF) {
      acc[nf] = mfma16t(aq, lds_frag_t(k_s, nf * 16, D, 0), acc[nf]);
      }
    }
#pragma unroll
    for (int nf = 0; nf < TMAXNF; ++nf) {
      if (nf < NF) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int lr = ((lane >> 4) << 2) + r;
        const int row = m0 + lr;
        const int col = nf * 16 + (lane & 15);
        float p = 0.f;
        if (row < L && col < len) {
          const int off = col - row + L - 1;
          const float s = acc[nf][r] + to_f32(bdw[(long)lr * 2 * Lpad + off]);
          p = __expf(s - lse_s[row]);
        }
        pt_s[(long)col * Lpad + row] = __float2bfloat16(p);
      }
      }
    }
  }
  __syncthreads();

  // ---- phase 2: dV = P^T dO ----
  for (int k0 = wid * 16; k0 < L; k0 += 64) {
    cfrag acc[2];
#pragma unroll
    for (int i = 0; i < 2; ++i) acc[i] = cfrag{0.f, 0.f, 0.f, 0.f};
    for (int kk = 0; kk < Lpad / 32; ++kk) {
      const bfrag ap = lds_frag_t(pt_s, k0, Lpad, kk * 32);
#pragma unroll
      for (int nd = 0; nd < 2; ++nd) {
        bfrag bo;
#pragma unroll
        for (int e = 0; e < 8; ++e)
          reinterpret_cast<bf16*>(&bo)[e] =
              do_s[(long)(kk * 32 + ((lane >> 4) << 3) + e) * D + nd * 16 +
                   (lane & 15)];
        acc[nd] = mfma16t(ap, bo, acc[nd]);
      }
    }
#pragma unroll
    for (int nd = 0; nd < 2; ++nd)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int krow = k0 + ((lane >> 4) << 2) + r;
        if (krow < L)
          dv[base + (long)krow * D + nd * 16 + (lane & 15)] =
              __float2bfloat16(acc[nd][r]);
      }
  }
  __syncthreads();

  // ---- phase 3: dP = dO V^T ; dS in place of P^T ----
  for (int m0 = wid * 16; m0 < L; m0 += 64) {
    cfrag acc[TMAXNF];
#pragma unroll
    for (int nf = 0; nf < TMAXNF; ++nf) acc[nf] = cfrag{0.f, 0.f, 0.f, 0.f};
    const bfrag ado = lds_frag_t(do_s, m0, D, 0);
#pragma unroll
    for (int nf = 0; nf < TMAXNF; ++nf) {
      if (nf < NF) {
      acc[nf] = mfma16t(ado, lds_frag_t(v_s, nf * 16, D, 0), acc[nf]);
      }
    }
#pragma unroll
    for (int nf = 0; nf < TMAXNF; ++nf) {
      if (nf < NF) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = m0 + ((lane >> 4) << 2) + r;
        const int col = nf * 16 + (lane & 15);
        const float p = to_f32(pt_s[(long)col * Lpad + row]);
        const float ds = p * (acc[nf][r] - delta_s[row]);  // unscaled attn
        pt_s[(long)col * Lpad + row] = __float2bfloat16(ds);
      }
      }
    }
  }
  __syncthreads();

  // ---- phase 4: dK = dS^T qu ----
  for (int k0 = wid * 16; k0 < L; k0 += 64) {
    cfrag acc[2];
#pragma unroll
    for (int i = 0; i < 2; ++i) acc[i] = cfrag{0.f, 0.f, 0.f, 0.f};
    for (int kk = 0; kk < Lpad / 32; ++kk) {
      const bfrag ads = lds_frag_t(pt_s, k0, Lpad, kk * 32);
#pragma unroll
      for (int nd = 0; nd < 2; ++nd) {
        bfrag bq;
#pragma unroll
        for (int e = 0; e < 8; ++e)
          reinterpret_cast<bf16*>(&bq)[e] =
              qu_s[(long)(kk * 32 + ((lane >> 4) << 3) + e) * D + nd * 16 +
                   (lane & 15)];
        acc[nd] = mfma16t(ads, bq, acc[nd]);
      }
    }
#pragma unroll
    for (int nd = 0; nd < 2; ++nd)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int krow = k0 + ((lane >> 4) << 2) + r;
        if (krow < L)
          dk[base + (long)krow * D + nd * 16 + (lane & 15)] =
              __float2bfloat16(acc[nd][r]);
      }
  }

  // ---- phase 5: dqu = dS K ; dBD scatter ; dqv = dBD R ----
  for (int m0 = wid * 16; m0 < L; m0 += 64) {
    // zero this wave's dBD tile
    for (int i = lane; i < 16 * 2 * Lpad / 8; i += WAVE)
      reinterpret_cast<s16x8*>(bdw)[i] = s16x8{};
    cfrag acc[2];
#pragma unroll
    for (int i = 0; i < 2; ++i) acc[i] = cfrag{0.f, 0.f, 0.f, 0.f};
    for (int kk = 0; kk < Lpad / 32; ++kk) {
      bfrag ads;
#pragma unroll
      for (int e = 0; e < 8; ++e)
        reinterpret_cast<bf16*>(&ads)[e] =
            pt_s[(long)(kk * 32 + ((lane >> 4) << 3) + e) * Lpad + m0 +
                 (lane & 15)];
#pragma unroll
      for (int nd = 0; nd < 2; ++nd) {
        // B = K[key, d] row-major -> per-element reads (not lds_frag_t)
        bfrag bk;
#pragma unroll
        for (int e = 0; e < 8; ++e)
          reinterpret_cast<bf16*>(&bk)[e] =
              k_s[(long)(kk * 32 + ((lane >> 4) << 3) + e) * D + nd * 16 +
                  (lane & 15)];
        acc[nd] = mfma16t(ads, bk, acc[nd]);
      }
    }
#pragma unroll
    for (int nd = 0; nd < 2; ++nd)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = m0 + ((lane >> 4) << 2) + r;
        if (row < L)
          dqu[base + (long)row * D + nd * 16 + (lane & 15)] =
              __float2bfloat16(acc[nd][r]);
      }
    // scatter dS -> dBD[lr][col - row + L - 1]
    for (int col = lane; col < L; col += WAVE) {
#pragma unroll
      for (int lr = 0; lr < 16; ++lr) {
        const int row = m0 + lr;
        if (row < L) {
        const int off = col - row + L - 1;
        bdw[(long)lr * 2 * Lpad + off] = pt_s[(long)col * Lpad + row];
        }
      }
    }
    // dqv tile = dBD @ R : [16, 2Lpad] x [2Lpad, D]
    cfrag vacc[2];
#pragma unroll
    for (int i = 0; i < 2; ++i) vacc[i] = cfrag{0.f, 0.f, 0.f, 0.f};
    for (int kk = 0; kk < 2 * Lpad / 32; ++kk) {
      const bfrag abd = lds_frag_t(bdw, 0, 2 * Lpad, kk * 32);
#pragma unroll
      for (int nd = 0; nd < 2; ++nd) {
        // B = R[m, d]: B[k=m][col=d] -> strided per-element reads
        bfrag br;
#pragma unroll
        for (int e = 0; e < 8; ++e)
          reinterpret_cast<bf16*>(&br)[e] =
              r_s[(long)(kk * 32 + ((lane >> 4) << 3) + e) * D + nd * 16 +
                  (lane & 15)];
        vacc[nd] = mfma16t(abd, br, vacc[nd]);
      }
    }
#pragma unroll
    for (int nd = 0; nd < 2; ++nd)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = m0 + ((lane >> 4) << 2) + r;
        if (row < L)
          dqv[base + (long)row * D + nd * 16 + (lane & 15)] =
              __float2bfloat16(vacc[nd][r]);
      }
  }
}

// ===================================================================== host
std::vector<at::Tensor> tener_attn_fwd(const at::Tensor& qu,
                                       const at::Tensor& qv,
                                       const at::Tensor& k,
                                       const at::Tensor& v,
                                       const at::Tensor& rel,
                                       const at::Tensor& lens) {
  CHECK_CUDA_CONTIG(qu);
  const int B = qu.size(0), H = qu.size(1), L = qu.size(2), D = qu.size(3);
  TORCH_CHECK(D == 32, "tener kernel: head dim must be padded to 32");
  const int Lpad = ((L + 31) / 32) * 32;
  TORCH_CHECK(Lpad <= TLPAD_MAX, "tener kernel: seq len > 160 unsupported");
  auto out = at::empty_like(qu);
  auto lse = at::empty({B, H, L}, qu.options().dtype(at::kFloat));
  const size_t smem =
      (size_t)(3 * Lpad * D + 2 * Lpad * D + 4 * 16 * 2 * Lpad +
               4 * 16 * Lpad) * sizeof(bf16);
  TORCH_CHECK(smem <= 160 * 1024, "tener fwd LDS overflow");
  hipLaunchKernelGGL(tener_fwd_kernel, dim3(B * H), dim3(256), smem,
                     cur_stream(qu), (const bf16*)qu.data_ptr(),
                     (const bf16*)qv.data_ptr(), (const bf16*)k.data_ptr(),
                     (const bf16*)v.data_ptr(), (const bf16*)rel.data_ptr(),
                     lens.data_ptr<int>(), (bf16*)out.data_ptr(),
                     lse.data_ptr<float>(), B, H, L, D, Lpad);
  HIP_CHECK_LAST();
  return {out, lse};
}

std::vector<at::Tensor> tener_attn_bwd(
    const at::Tensor& dout, const at::Tensor& qu, const at::Tensor& qv,
    const at::Tensor& k, const at::Tensor& v, const at::Tensor& rel,
    const at::Tensor& o, const at::Tensor& lse, const at::Tensor& lens) {
  CHECK_CUDA_CONTIG(qu);
  const int B = qu.size(0), H = qu.size(1), L = qu.size(2), D = qu.size(3);
  TORCH_CHECK(D == 32, "tener kernel: head dim must be padded to 32");
  const int Lpad = ((L + 31) / 32) * 32;
  TORCH_CHECK(Lpad <= TLPAD_MAX, "tener kernel: seq len > 160 unsupported");
  auto dqu = at::empty_like(qu);
  auto dqv = at::empty_like(qu);
  auto dk = at::empty_like(k);
  auto dv = at::empty_like(v);
  const size_t smem =
      (size_t)(3 * Lpad * D + 2 * Lpad * D + 4 * 16 * 2 * Lpad + Lpad * D +
               (long)Lpad * Lpad) * sizeof(bf16) + 2 * Lpad * sizeof(float);
  TORCH_CHECK(smem <= 160 * 1024, "tener bwd LDS overflow: ", smem);
  hipLaunchKernelGGL(tener_bwd_kernel, dim3(B * H), dim3(256), smem,
                     cur_stream(qu), (const bf16*)dout.data_ptr(),
                     (const bf16*)qu.data_ptr(), (const bf16*)qv.data_ptr(),
                     (const bf16*)k.data_ptr(), (const bf16*)v.data_ptr(),
                     (const bf16*)rel.data_ptr(), (const bf16*)o.data_ptr(),
                     lse.data_ptr<float>(), lens.data_ptr<int>(),
                     (bf16*)dqu.data_ptr(), (bf16*)dqv.data_ptr(),
                     (bf16*)dk.data_ptr(), (bf16*)dv.data_ptr(), B, H, L, D,
                     Lpad);
  HIP_CHECK_LAST();
  return {dqu, dqv, dk, dv};
}
