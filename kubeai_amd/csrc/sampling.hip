// Token sampling kernels.
//
// greedy_sample:  argmax over the vocab row.
// gumbel_sample:  argmax(logits/T + Gumbel noise) — exact temperature
//   sampling without softmax or sort (one reduction-shaped kernel). The RNG
//   is the counter-based splitmix64 hash in common.h, spec'd identically in
//   ops/ref.py so kernel and reference produce bit-identical samples.
// Tie-break: lowest index wins (matches torch.argmax).
#include <torch/extension.h>

#include "common.h"

namespace {

constexpr int kBlock = 256;

template <bool kGumbel>
__global__ void sample_kernel(int64_t* __restrict__ out,       // [B]
                              const float* __restrict__ logits,  // [B, V]
                              const float* __restrict__ temperature,  // [B]
                              const int64_t* __restrict__ seeds,      // [B]
                              const int64_t step, const int V) {
  const int b = blockIdx.x;
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
  for (int v = threadIdx.x; v < V; v += kBlock) {
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

  // wave reduce (value, index) with lowest-index tie-break
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
    out[b] = best_idx;
  }
}

}  // namespace

void greedy_sample(torch::Tensor out, torch::Tensor logits) {
  TORCH_CHECK(logits.is_contiguous());
  TORCH_CHECK(logits.scalar_type() == torch::kFloat32,
              "sampling expects fp32 logits");
  const int B = logits.size(0), V = logits.size(1);
  if (B == 0) return;
  hipLaunchKernelGGL((sample_kernel<false>), dim3(B), dim3(kBlock), 0,
                     c10::hip::getCurrentHIPStream().stream(),
                     out.data_ptr<int64_t>(), logits.data_ptr<float>(),
                     nullptr, nullptr, 0, V);
  HIP_CHECK_KERNEL();
}

void gumbel_sample(torch::Tensor out, torch::Tensor logits,
                   torch::Tensor temperature, torch::Tensor seeds,
                   int64_t step) {
  TORCH_CHECK(logits.is_contiguous());
  TORCH_CHECK(logits.scalar_type() == torch::kFloat32);
  TORCH_CHECK(temperature.scalar_type() == torch::kFloat32);
  TORCH_CHECK(seeds.scalar_type() == torch::kInt64);
  const int B = logits.size(0), V = logits.size(1);
  if (B == 0) return;
  hipLaunchKernelGGL((sample_kernel<true>), dim3(B), dim3(kBlock), 0,
                     c10::hip::getCurrentHIPStream().stream(),
                     out.data_ptr<int64_t>(), logits.data_ptr<float>(),
                     temperature.data_ptr<float>(),
                     seeds.data_ptr<int64_t>(), step, V);
  HIP_CHECK_KERNEL();
}
