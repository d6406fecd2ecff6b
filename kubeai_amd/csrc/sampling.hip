// Token sampling kernels (two-stage argmax reduction).
//
// greedy_sample:  argmax over the vocab row.
// gumbel_sample:  argmax(logits/T + Gumbel noise) — exact temperature
//   sampling without softmax or sort. RNG = counter-based splitmix64 hash
//   (common.h), spec-identical to ops/ref.py.
// Tie-break: lowest index wins (matches torch.argmax).
//
// Stage 1 splits each vocab row into kSplits chunks (B*kSplits workgroups
// keep all 256 CUs busy even at decode batch ~32); stage 2 is one wave per
// row reducing the partials.
#include <torch/extension.h>

#include "common.h"

namespace {

constexpr int kBlock = 256;
constexpr int kSplits = 32;

template <bool kGumbel>
__global__ void sample_stage1(
    float* __restrict__ part_val,          // [B, kSplits]
    int32_t* __restrict__ part_idx,        // [B, kSplits]
    const float* __restrict__ logits,      // [B, V]
    const float* __restrict__ temperature,  // [B]
    const int64_t* __restrict__ seeds,      // [B]
    const int64_t step, const int V) {
  const int b = blockIdx.x;
  const int split = blockIdx.y;
  const int chunk = (V + kSplits - 1) / kSplits;
  const int lo = split * chunk;
  const int hi = min(V, lo + chunk);
  const float* row = logits + (int64_t)b * V;
  float temp = 1.0f;
  uint64_t key = 0;
  bool greedy = !kGumbel;
  if constexpr (kGumbel) {
    temp = temperature[b];
    if (temp <= 0.f) greedy = true;
    key = (uint64_t)seeds[b] * 1000003ull + (uint64_t)step;
  }

  float best = -INFINITY;
  int best_idx = V;
  for (int v = lo + threadIdx.x; v < hi; v += kBlock) {
    float val = row[v];
    if (!greedy) {
      float u = hash_uniform(key, (uint64_t)v);
      val = val / temp + (-__logf(-__logf(u)));
    }
    if (val > best || (val == best && v < best_idx)) {
      best = val;
      best_idx = v;
    }
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    float ov = __shfl_xor(best, off, 64);
    int oi = __shfl_xor(best_idx, off, 64);
    if (ov > best || (ov == best && oi < best_idx)) {
      best = ov;
      best_idx = oi;
    }
  }
  __shared__ float sval[4];
  __shared__ int sidx[4];
  const int wave = threadIdx.x / WAVE_SIZE;
  if (threadIdx.x % WAVE_SIZE == 0) {
    sval[wave] = best;
    sidx[wave] = best_idx;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
#pragma unroll
    for (int w = 1; w < kBlock / WAVE_SIZE; ++w) {
      if (sval[w] > best || (sval[w] == best && sidx[w] < best_idx)) {
        best = sval[w];
        best_idx = sidx[w];
      }
    }
    part_val[b * kSplits + split] = best;
    part_idx[b * kSplits + split] = best_idx;
  }
}

__global__ void sample_stage2(int64_t* __restrict__ out,
                              const float* __restrict__ part_val,
                              const int32_t* __restrict__ part_idx) {
  const int b = blockIdx.x;
  const int lane = threadIdx.x;
  float best = (lane < kSplits) ? part_val[b * kSplits + lane] : -INFINITY;
  int best_idx = (lane < kSplits) ? part_idx[b * kSplits + lane] : INT32_MAX;
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    float ov = __shfl_xor(best, off, 64);
    int oi = __shfl_xor(best_idx, off, 64);
    if (ov > best || (ov == best && oi < best_idx)) {
      best = ov;
      best_idx = oi;
    }
  }
  if (lane == 0) out[b] = best_idx;
}

template <bool kGumbel>
void launch_sample(torch::Tensor out, torch::Tensor logits,
                   const float* temp_ptr, const int64_t* seed_ptr,
                   int64_t step) {
  const int B = logits.size(0), V = logits.size(1);
  if (B == 0) return;
  auto opts = torch::TensorOptions().device(logits.device());
  auto part_val = torch::empty({B, kSplits}, opts.dtype(torch::kFloat32));
  auto part_idx = torch::empty({B, kSplits}, opts.dtype(torch::kInt32));
  auto stream = c10::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL((sample_stage1<kGumbel>), dim3(B, kSplits), dim3(kBlock),
                     0, stream, part_val.data_ptr<float>(),
                     part_idx.data_ptr<int32_t>(), logits.data_ptr<float>(),
                     temp_ptr, seed_ptr, step, V);
  HIP_CHECK_KERNEL();
  hipLaunchKernelGGL(sample_stage2, dim3(B), dim3(WAVE_SIZE), 0, stream,
                     out.data_ptr<int64_t>(), part_val.data_ptr<float>(),
                     part_idx.data_ptr<int32_t>());
  HIP_CHECK_KERNEL();
}

}  // namespace

void greedy_sample(torch::Tensor out, torch::Tensor logits) {
  TORCH_CHECK(logits.is_contiguous());
  TORCH_CHECK(logits.scalar_type() == torch::kFloat32,
              "sampling expects fp32 logits");
  launch_sample<false>(out, logits, nullptr, nullptr, 0);
}

void gumbel_sample(torch::Tensor out, torch::Tensor logits,
                   torch::Tensor temperature, torch::Tensor seeds,
                   int64_t step) {
  TORCH_CHECK(logits.is_contiguous());
  TORCH_CHECK(logits.scalar_type() == torch::kFloat32);
  TORCH_CHECK(temperature.scalar_type() == torch::kFloat32);
  TORCH_CHECK(seeds.scalar_type() == torch::kInt64);
  launch_sample<true>(out, logits, temperature.data_ptr<float>(),
                      seeds.data_ptr<int64_t>(), step);
}

// ---------------------------------------------------------------------
// Rejection-based nucleus sampling support (runner._sample_topk_topp):
// one pass computes per-row softmax stats; per candidate draw another
// single pass decides membership (probability mass strictly above the
// sampled logit < top_p, and rank < top_k) — no [S,V] intermediates.

namespace {

__global__ void nucleus_stats_kernel(
    float* __restrict__ m_out, float* __restrict__ z_out,
    const float* __restrict__ logits, const float* __restrict__ temps,
    const int V) {
  const int row = blockIdx.x;
  const float* x = logits + (int64_t)row * V;
  const float t = fmaxf(temps[row], 1e-6f);
  __shared__ float red[16];
  float m = -INFINITY;
  for (int i = threadIdx.x; i < V; i += blockDim.x) m = fmaxf(m, x[i]);
  m = wave_reduce_max(m);
  const int wave = threadIdx.x / WAVE_SIZE, lane = threadIdx.x % WAVE_SIZE;
  const int nw = blockDim.x / WAVE_SIZE;
  if (lane == 0) red[wave] = m;
  __syncthreads();
  float mm = -INFINITY;
  for (int w = 0; w < nw; ++w) mm = fmaxf(mm, red[w]);
  __syncthreads();
  float z = 0.f;
  for (int i = threadIdx.x; i < V; i += blockDim.x)
    z += __expf((x[i] - mm) / t);
  z = block_reduce_sum(z, red);
  if (threadIdx.x == 0) {
    m_out[row] = mm;
    z_out[row] = z;
  }
}

__global__ void nucleus_accept_kernel(
    unsigned char* __restrict__ ok,      // [S]
    const float* __restrict__ logits,    // [S, V]
    const int64_t* __restrict__ cand,    // [S]
    const float* __restrict__ m_in, const float* __restrict__ z_in,
    const float* __restrict__ temps, const float* __restrict__ top_ps,
    const int32_t* __restrict__ top_ks, const int V) {
  const int row = blockIdx.x;
  const float t0 = temps[row];
  if (t0 <= 0.f) {  // greedy rows: argmax is always in the nucleus
    if (threadIdx.x == 0) ok[row] = 1;
    return;
  }
  const float* x = logits + (int64_t)row * V;
  const float lt = x[cand[row]];
  const float t = fmaxf(t0, 1e-6f);
  const float m = m_in[row], z = z_in[row];
  __shared__ float red[16];
  float mass = 0.f;
  int cnt = 0;
  for (int i = threadIdx.x; i < V; i += blockDim.x) {
    const float xi = x[i];
    if (xi > lt) {
      mass += __expf((xi - m) / t);
      ++cnt;
    }
  }
  mass = block_reduce_sum(mass, red);
  __syncthreads();
  float cntf = block_reduce_sum((float)cnt, red);
  if (threadIdx.x == 0) {
    const float mass_above = mass / z;
    const int k = top_ks[row];
    ok[row] = (mass_above < top_ps[row] && (k <= 0 || (int)cntf < k)) ? 1 : 0;
  }
}

}  // namespace

void nucleus_stats(torch::Tensor m, torch::Tensor z, torch::Tensor logits,
                   torch::Tensor temps) {
  TORCH_CHECK(logits.is_contiguous() &&
              logits.scalar_type() == torch::kFloat32);
  const int S = logits.size(0), V = logits.size(1);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(nucleus_stats_kernel, dim3(S), dim3(512), 0, stream,
                     m.data_ptr<float>(), z.data_ptr<float>(),
                     logits.data_ptr<float>(), temps.data_ptr<float>(), V);
  HIP_CHECK_KERNEL();
}

void nucleus_accept(torch::Tensor ok, torch::Tensor logits,
                    torch::Tensor cand, torch::Tensor m, torch::Tensor z,
                    torch::Tensor temps, torch::Tensor top_ps,
                    torch::Tensor top_ks) {
  TORCH_CHECK(logits.is_contiguous() &&
              logits.scalar_type() == torch::kFloat32);
  TORCH_CHECK(cand.scalar_type() == torch::kInt64);
  TORCH_CHECK(top_ks.scalar_type() == torch::kInt32);
  const int S = logits.size(0), V = logits.size(1);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(nucleus_accept_kernel, dim3(S), dim3(512), 0, stream,
                     ok.data_ptr<unsigned char>(),
                     logits.data_ptr<float>(), cand.data_ptr<int64_t>(),
                     m.data_ptr<float>(), z.data_ptr<float>(),
                     temps.data_ptr<float>(), top_ps.data_ptr<float>(),
                     top_ks.data_ptr<int32_t>(), V);
  HIP_CHECK_KERNEL();
}
