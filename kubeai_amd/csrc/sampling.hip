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
