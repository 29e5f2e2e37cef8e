// Linear-chain CRF: forward-backward (loss + analytic grads in one pass)
// and Viterbi decode. SURVEY.md K5/K6: T is tiny (<= 20), sequences are
// independent -> one wave per sequence, lane j owns tag j, transition
// matrix in LDS, alphas staged per-timestep in LDS for the backward pass.
#include "common.h"

#define TMAX 20
#define NEG (-1e30f)

// one wave per sequence; blockDim = 4 waves
__global__ void crf_fwd_kernel(const float* __restrict__ emis,  // [B,L,T]
                               const int* __restrict__ tags,    // [B,L]
                               const int* __restrict__ lens,    // [B]
                               const float* __restrict__ trans, // [T,T]
                               float* __restrict__ ll,          // [B]
                               float* __restrict__ demis,       // [B,L,T] zeroed
                               float* __restrict__ dtrans,      // [B,T,T] zeroed
                               int B, int L, int T) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  float* tr_s = reinterpret_cast<float*>(smem_raw);            // [T*T]
  float* alpha_all = tr_s + TMAX * TMAX;  // [waves][L][T]
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int nw = blockDim.x / WAVE;
  for (int i = threadIdx.x; i < T * T; i += blockDim.x) tr_s[i] = trans[i];
  __syncthreads();

  const int b = blockIdx.x * nw + wid;
  if (b >= B) return;
  const int n = lens[b];
  if (n <= 0) return;
  const float* e = emis + (long)b * L * T;
  const int* tg = tags + (long)b * L;
  float* alpha = alpha_all + (long)wid * L * T;
  const int j = lane;  // tag owned by this lane (j < T active)
  const bool act = j < T;

  // ---- forward pass: alpha[t][:] in LDS ----
  if (act) alpha[j] = e[j];
  for (int t = 1; t < n; ++t) {
    float m = NEG;
    if (act) {
      for (int i = 0; i < T; ++i)
        m = fmaxf(m, alpha[(t - 1) * T + i] + tr_s[i * T + j]);
      float s = 0.f;
      for (int i = 0; i < T; ++i)
        s += __expf(alpha[(t - 1) * T + i] + tr_s[i * T + j] - m);
      alpha[t * T + j] = m + __logf(s) + e[t * T + j];
    }
  }
  // logZ = lse over lanes j < T of alpha[n-1]
  float av = act ? alpha[(n - 1) * T + j] : NEG;
  const float mz = wave_reduce_max(av);
  const float sz = wave_reduce_sum(act ? __expf(av - mz) : 0.f);
  const float logZ = mz + __logf(sz);

  // ---- gold score (lane 0, serial — n steps of scalar work) ----
  if (lane == 0) {
    float sc = e[tg[0]];
    for (int t = 1; t < n; ++t)
      sc += tr_s[tg[t - 1] * T + tg[t]] + e[t * T + tg[t]];
    ll[b] = sc - logZ;
  }

  // ---- backward pass + grads ----
  // beta held in registers per lane; expected transition counts
  // accumulated per lane i over its row.
  float* de = demis + (long)b * L * T;
  float beta = 0.f;  // beta[n-1][j] = 0
  if (act) {
    const float marg = __expf(alpha[(n - 1) * T + j] + beta - logZ);
    de[(n - 1) * T + j] = ((j == tg[n - 1]) ? 1.f : 0.f) - marg;
  }
  float exp_row[TMAX];  // lane i: sum_t P(y_t=i, y_{t+1}=j)
#pragma unroll
  for (int q = 0; q < TMAX; ++q) exp_row[q] = 0.f;
  // LDS row buffer for (e[t+1][jj] + beta[t+1][jj]) — reuse the tail of
  // this wave's alpha buffer? alpha is still needed; use a separate slot:
  // stash in alpha[(L-1)*T .. ] is unsafe; instead broadcast via shfl.
  for (int t = n - 2; t >= 0; --t) {
    // cur[jj] = e[t+1][jj] + beta[t+1][jj]; each lane jj<T holds its own;
    // every lane needs all -> read via __shfl
    const float mine = act ? e[(t + 1) * T + j] + beta : NEG;
    float m = NEG, s = 0.f;
    float row[TMAX];  // trans[i=lane][jj] + cur[jj]
#pragma unroll
    for (int jj = 0; jj < TMAX; ++jj) {
      if (jj < T) {
      const float cur = __shfl(mine, jj);
      const float v = (act ? tr_s[j * T + jj] : NEG) + cur;
      row[jj] = v;
      m = fmaxf(m, v);
      }
    }
#pragma unroll
    for (int jj = 0; jj < TMAX; ++jj) {
      if (jj < T) {
      s += __expf(row[jj] - m);
      }
    }
    const float beta_t = act ? m + __logf(s) : NEG;  // beta[t][i=lane]
    // expected pairwise counts: P(y_t=i, y_{t+1}=jj)
    if (act) {
      const float a_ti = alpha[t * T + j];  // lane = i here
#pragma unroll
      for (int jj = 0; jj < TMAX; ++jj) {
        if (jj < T) {
        exp_row[jj] += __expf(a_ti + row[jj] - logZ);
        }
      }
      // token marginal at t for tag i=lane
      const float marg = __expf(a_ti + beta_t - logZ);
      de[t * T + j] = ((j == tg[t]) ? 1.f : 0.f) - marg;
    }
    beta = beta_t;
  }
  // gold pair counts minus expected -> dtrans[b]
  if (act) {
    float gold[TMAX];
#pragma unroll
    for (int q = 0; q < TMAX; ++q) gold[q] = 0.f;
    for (int t = 1; t < n; ++t) {
      if (tg[t - 1] == j) {
#pragma unroll
        for (int jj = 0; jj < TMAX; ++jj) {
          if (jj < T) {
          if (tg[t] == jj) gold[jj] += 1.f;
          }
        }
      }
    }
    float* dt = dtrans + ((long)b * T + j) * T;
#pragma unroll
    for (int jj = 0; jj < TMAX; ++jj) {
      if (jj < T) {
      dt[jj] = gold[jj] - exp_row[jj];
      }
    }
  }
}

// ---------------------------------------------------------------------
// 4-sequences-per-wave variant (T <= 16, the NER/CWS regime): lane
// group g = lane>>4 owns sequence 4*wid+g, lane j = lane&15 owns tag j.
// Quadruples the active lanes of the latency-bound scan (the one-wave
// kernel ran 10/64 lanes; standalone kernel time 424 us at bs64 L128).
// All shfl traffic stays inside the 16-lane group (groups share one
// sequence, so loop trip counts are group-uniform).
// ---------------------------------------------------------------------
#define T16 16

__global__ void crf_fwd_kernel_x4(const float* __restrict__ emis,
                                  const int* __restrict__ tags,
                                  const int* __restrict__ lens,
                                  const float* __restrict__ trans,
                                  float* __restrict__ ll,
                                  float* __restrict__ demis,   // zeroed
                                  float* __restrict__ dtrans,  // zeroed
                                  int B, int L, int T) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  float* tr_s = reinterpret_cast<float*>(smem_raw);            // [T*T]
  float* alpha_all = tr_s + T16 * T16;  // [waves*4][L][T]
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int nw = blockDim.x / WAVE;
  const int g = lane >> 4;
  const int base = lane & 48;          // group's first lane
  for (int i = threadIdx.x; i < T * T; i += blockDim.x) tr_s[i] = trans[i];
  __syncthreads();

  const int b = blockIdx.x * nw * 4 + wid * 4 + g;
  if (b >= B) return;
  const int n = lens[b];
  if (n <= 0) return;
  const float* e = emis + (long)b * L * T;
  const int* tg = tags + (long)b * L;
  float* alpha = alpha_all + (long)(wid * 4 + g) * L * T;
  const int j = lane & 15;
  const bool act = j < T;

  if (act) alpha[j] = e[j];
  for (int t = 1; t < n; ++t) {
    float m = NEG;
    if (act) {
      for (int i = 0; i < T; ++i)
        m = fmaxf(m, alpha[(t - 1) * T + i] + tr_s[i * T + j]);
      float s = 0.f;
      for (int i = 0; i < T; ++i)
        s += __expf(alpha[(t - 1) * T + i] + tr_s[i * T + j] - m);
      alpha[t * T + j] = m + __logf(s) + e[t * T + j];
    }
  }
  float av = act ? alpha[(n - 1) * T + j] : NEG;
  const float mz = group16_reduce_max(av);
  const float sz = group16_reduce_sum(act ? __expf(av - mz) : 0.f);
  const float logZ = mz + __logf(sz);

  if (j == 0) {
    float sc = e[tg[0]];
    for (int t = 1; t < n; ++t)
      sc += tr_s[tg[t - 1] * T + tg[t]] + e[t * T + tg[t]];
    ll[b] = sc - logZ;
  }

  float* de = demis + (long)b * L * T;
  float beta = 0.f;
  if (act) {
    const float marg = __expf(alpha[(n - 1) * T + j] + beta - logZ);
    de[(n - 1) * T + j] = ((j == tg[n - 1]) ? 1.f : 0.f) - marg;
  }
  float exp_row[T16];
#pragma unroll
  for (int q = 0; q < T16; ++q) exp_row[q] = 0.f;
  for (int t = n - 2; t >= 0; --t) {
    const float mine = act ? e[(t + 1) * T + j] + beta : NEG;
    float m = NEG, s = 0.f;
    float row[T16];
#pragma unroll
    for (int jj = 0; jj < T16; ++jj) {
      if (jj < T) {
        const float cur = __shfl(mine, base + jj);
        const float v = (act ? tr_s[j * T + jj] : NEG) + cur;
        row[jj] = v;
        m = fmaxf(m, v);
      }
    }
#pragma unroll
    for (int jj = 0; jj < T16; ++jj)
      if (jj < T) s += __expf(row[jj] - m);
    const float beta_t = act ? m + __logf(s) : NEG;
    if (act) {
      const float a_ti = alpha[t * T + j];
#pragma unroll
      for (int jj = 0; jj < T16; ++jj)
        if (jj < T) exp_row[jj] += __expf(a_ti + row[jj] - logZ);
      const float marg = __expf(a_ti + beta_t - logZ);
      de[t * T + j] = ((j == tg[t]) ? 1.f : 0.f) - marg;
    }
    beta = beta_t;
  }
  if (act) {
    float gold[T16];
#pragma unroll
    for (int q = 0; q < T16; ++q) gold[q] = 0.f;
    for (int t = 1; t < n; ++t) {
      if (tg[t - 1] == j) {
#pragma unroll
        for (int jj = 0; jj < T16; ++jj)
          if (jj < T && tg[t] == jj) gold[jj] += 1.f;
      }
    }
    float* dt = dtrans + ((long)b * T + j) * T;
#pragma unroll
    for (int jj = 0; jj < T16; ++jj)
      if (jj < T) dt[jj] = gold[jj] - exp_row[jj];
  }
}

__global__ void crf_viterbi_kernel_x4(const float* __restrict__ emis,
                                      const int* __restrict__ lens,
                                      const float* __restrict__ trans,
                                      int* __restrict__ pred,  // zeroed
                                      int B, int L, int T) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  float* tr_s = reinterpret_cast<float*>(smem_raw);
  const int nw = blockDim.x / WAVE;
  const size_t bp_stride = (size_t)L * T16;
  char* bp_all = smem_raw + T16 * T16 * sizeof(float);
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int g = lane >> 4;
  const int base = lane & 48;
  for (int i = threadIdx.x; i < T * T; i += blockDim.x) tr_s[i] = trans[i];
  __syncthreads();

  const int b = blockIdx.x * nw * 4 + wid * 4 + g;
  if (b >= B) return;
  const int n = lens[b];
  if (n <= 0) return;
  const float* e = emis + (long)b * L * T;
  char* bp = bp_all + (size_t)(wid * 4 + g) * bp_stride;
  const int j = lane & 15;
  const bool act = j < T;
  float delta = act ? e[j] : NEG;
  for (int t = 1; t < n; ++t) {
    float best = NEG;
    int arg = 0;
    for (int i = 0; i < T; ++i) {
      const float dprev = __shfl(delta, base + i);
      const float v = dprev + (act ? tr_s[i * T + j] : NEG);
      if (v > best) {
        best = v;
        arg = i;
      }
    }
    delta = act ? best + e[t * T + j] : NEG;
    if (act) bp[t * T16 + j] = (char)arg;
  }
  float bv = delta;
  int barg = act ? j : 0;
#pragma unroll
  for (int off = 8; off > 0; off >>= 1) {   // within the 16-lane group
    const float ov = __shfl_xor(bv, off);
    const int oa = __shfl_xor(barg, off);
    if (ov > bv) {
      bv = ov;
      barg = oa;
    }
  }
  if (j == 0) {
    int* pr = pred + (long)b * L;
    int cur = min(max(barg, 0), T - 1);
    pr[n - 1] = cur;
    for (int t = n - 2; t >= 0; --t) {
      cur = min(max((int)bp[(t + 1) * T16 + cur], 0), T - 1);
      pr[t] = cur;
    }
  }
}

// Viterbi: same layout; backpointers in LDS (int8), lane-0 backtrace.
__global__ void crf_viterbi_kernel(const float* __restrict__ emis,
                                   const int* __restrict__ lens,
                                   const float* __restrict__ trans,
                                   int* __restrict__ pred,  // [B,L] zeroed
                                   int B, int L, int T) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  float* tr_s = reinterpret_cast<float*>(smem_raw);          // [TMAX*TMAX]
  const int nw = blockDim.x / WAVE;
  // per-wave: delta row [T] floats + bp [L][T] int8 (rounded to 16)
  const size_t bp_stride = (size_t)L * TMAX;
  char* bp_all = smem_raw + TMAX * TMAX * sizeof(float);
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  for (int i = threadIdx.x; i < T * T; i += blockDim.x) tr_s[i] = trans[i];
  __syncthreads();

  const int b = blockIdx.x * nw + wid;
  if (b >= B) return;
  const int n = lens[b];
  if (n <= 0) return;
  const float* e = emis + (long)b * L * T;
  char* bp = bp_all + (size_t)wid * bp_stride;
  const int j = lane;
  const bool act = j < T;
  float delta = act ? e[j] : NEG;  // delta[0][j]
  for (int t = 1; t < n; ++t) {
    // all lanes need delta[t-1][i] -> shfl broadcast
    float best = NEG;
    int arg = 0;
    for (int i = 0; i < T; ++i) {
      const float dprev = __shfl(delta, i);
      const float v = dprev + (act ? tr_s[i * T + j] : NEG);
      if (v > best) {
        best = v;
        arg = i;
      }
    }
    delta = act ? best + e[t * T + j] : NEG;
    if (act) bp[t * TMAX + j] = (char)arg;
  }
  // argmax over lanes
  float bv = delta;
  int barg = act ? j : 0;
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    const float ov = __shfl_xor(bv, off);
    const int oa = __shfl_xor(barg, off);
    if (ov > bv) {
      bv = ov;
      barg = oa;
    }
  }
  if (lane == 0) {
    int* pr = pred + (long)b * L;
    // clamp defensively: non-finite emissions (diverged training) must
    // not turn into out-of-bounds backtraces
    int cur = min(max(barg, 0), T - 1);
    pr[n - 1] = cur;
    for (int t = n - 2; t >= 0; --t) {
      cur = min(max((int)bp[(t + 1) * TMAX + cur], 0), T - 1);
      pr[t] = cur;
    }
  }
}

// ===================================================================== host
std::vector<at::Tensor> crf_fwd(const at::Tensor& emissions,
                                const at::Tensor& tags, const at::Tensor& lens,
                                const at::Tensor& trans) {
  CHECK_CUDA_CONTIG(emissions);
  const int B = emissions.size(0), L = emissions.size(1), T = emissions.size(2);
  TORCH_CHECK(T <= TMAX, "CRF: label_size > ", TMAX);
  auto ll = at::empty({B}, emissions.options());
  auto demis = at::zeros_like(emissions);
  auto dtrans = at::zeros({B, T, T}, emissions.options());
  const int nw = 4;
  const size_t smem4 = T16 * T16 * sizeof(float) +
                       (size_t)nw * 4 * L * T * sizeof(float);
  if (T <= T16 && smem4 <= 160 * 1024) {
    // 4 sequences per wave (16 per block): 4x the active lanes
    hipLaunchKernelGGL(crf_fwd_kernel_x4, dim3((B + nw * 4 - 1) / (nw * 4)),
                       dim3(nw * WAVE), smem4, cur_stream(emissions),
                       emissions.data_ptr<float>(), tags.data_ptr<int>(),
                       lens.data_ptr<int>(), trans.data_ptr<float>(),
                       ll.data_ptr<float>(), demis.data_ptr<float>(),
                       dtrans.data_ptr<float>(), B, L, T);
    HIP_CHECK_LAST();
    return {ll, demis, dtrans};
  }
  const size_t smem = TMAX * TMAX * sizeof(float) +
                      (size_t)nw * L * T * sizeof(float);
  TORCH_CHECK(smem <= 160 * 1024, "CRF fwd: LDS overflow (L*T too big)");
  hipLaunchKernelGGL(crf_fwd_kernel, dim3((B + nw - 1) / nw), dim3(nw * WAVE),
                     smem, cur_stream(emissions),
                     emissions.data_ptr<float>(), tags.data_ptr<int>(),
                     lens.data_ptr<int>(), trans.data_ptr<float>(),
                     ll.data_ptr<float>(), demis.data_ptr<float>(),
                     dtrans.data_ptr<float>(), B, L, T);
  HIP_CHECK_LAST();
  return {ll, demis, dtrans};
}

at::Tensor crf_viterbi(const at::Tensor& emissions, const at::Tensor& lens,
                       const at::Tensor& trans) {
  CHECK_CUDA_CONTIG(emissions);
  const int B = emissions.size(0), L = emissions.size(1), T = emissions.size(2);
  TORCH_CHECK(T <= TMAX, "CRF: label_size > ", TMAX);
  auto pred = at::zeros({B, L}, emissions.options().dtype(at::kInt));
  const int nw = 4;
  const size_t smem4 = T16 * T16 * sizeof(float) + (size_t)nw * 4 * L * T16;
  if (T <= T16 && smem4 <= 160 * 1024) {
    hipLaunchKernelGGL(crf_viterbi_kernel_x4,
                       dim3((B + nw * 4 - 1) / (nw * 4)), dim3(nw * WAVE),
                       smem4, cur_stream(emissions),
                       emissions.data_ptr<float>(), lens.data_ptr<int>(),
                       trans.data_ptr<float>(), pred.data_ptr<int>(), B, L, T);
    HIP_CHECK_LAST();
    return pred;
  }
  const size_t smem = TMAX * TMAX * sizeof(float) + (size_t)nw * L * TMAX;
  TORCH_CHECK(smem <= 160 * 1024, "CRF viterbi: LDS overflow");
  hipLaunchKernelGGL(crf_viterbi_kernel, dim3((B + nw - 1) / nw),
                     dim3(nw * WAVE), smem, cur_stream(emissions),
                     emissions.data_ptr<float>(), lens.data_ptr<int>(),
                     trans.data_ptr<float>(), pred.data_ptr<int>(), B, L, T);
  HIP_CHECK_LAST();
  return pred;
}
