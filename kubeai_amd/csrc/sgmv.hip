// SGMV (segmented gather matrix-vector) — multi-LoRA batched apply.
//
//   y[idx[i]] += B (A x[idx[i]]) * scale        for one adapter's segment
//
// The host launches one call per adapter present in the batch (few);
// token gathering rides the idx tensor. Shapes: A [R, H] (shrink),
// B [out, R] (expand), LoRA rank R <= 64.
//
// Design: workgroup per gathered token; phase 1 computes the R-vector
// v = A x (wave-split rows, lane-split H, butterfly reduce) into LDS;
// phase 2 streams B once per token with y[t, o] += dot(B[o, :], v)
// (v broadcast from LDS). A and B are small (<= ~350 KB combined) and
// stay L2-resident across the segment's workgroups.
#include <torch/extension.h>

#include "common.h"

namespace {

constexpr int kBlock = 256;
constexpr int kMaxR = 64;

__global__ void sgmv_kernel(
    ushort* __restrict__ y,        // [T, out] bf16 (+=)
    const ushort* __restrict__ x,  // [T, H] bf16
    const ushort* __restrict__ A,  // [R, H] bf16
    const ushort* __restrict__ B,  // [out, R] bf16
    const int64_t* __restrict__ idx,  // [n] gathered token rows
    const float scale, const int H, const int R, const int out,
    const int64_t x_stride, const int64_t y_stride) {
  const int64_t t = idx[blockIdx.x];
  const int wave = threadIdx.x / WAVE_SIZE;
  const int lane = threadIdx.x % WAVE_SIZE;
  const ushort8* xrow =
      reinterpret_cast<const ushort8*>(x + t * x_stride);

  __shared__ float v[kMaxR];

  // phase 1: v[r] = dot(A[r], x[t]) — waves split rows, lanes split H
  for (int r = wave; r < R; r += kBlock / WAVE_SIZE) {
    const ushort8* arow = reinterpret_cast<const ushort8*>(A + (int64_t)r * H);
    float acc = 0.f;
    for (int i = lane; i < H / 8; i += WAVE_SIZE) {
      const ushort8 xa = xrow[i];
      const ushort8 aa = arow[i];
#pragma unroll
      for (int j = 0; j < 8; ++j)
        acc = fmaf(bf16_to_f32(xa[j]), bf16_to_f32(aa[j]), acc);
    }
    acc = wave_reduce_sum(acc);
    if (lane == 0) v[r] = acc * scale;
  }
  __syncthreads();

  // phase 2: y[t, o] += dot(B[o, :R], v) — threads stride output features
  ushort* yrow = y + t * y_stride;
  for (int o = threadIdx.x; o < out; o += kBlock) {
    const ushort* brow = B + (int64_t)o * R;
    float acc = 0.f;
    for (int r = 0; r < R; ++r)
      acc = fmaf(bf16_to_f32(brow[r]), v[r], acc);
    yrow[o] = f32_to_bf16(bf16_to_f32(yrow[o]) + acc);
  }
}

}  // namespace

void sgmv(torch::Tensor y, torch::Tensor x, torch::Tensor A, torch::Tensor B,
          torch::Tensor idx, double scale) {
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(A.is_contiguous() && B.is_contiguous());
  TORCH_CHECK(idx.scalar_type() == torch::kInt64);
  const int H = x.size(1), R = A.size(0), out = B.size(0);
  TORCH_CHECK(A.size(1) == H && B.size(1) == R && R <= kMaxR);
  TORCH_CHECK(H % 8 == 0);
  TORCH_CHECK(x.stride(1) == 1 && y.stride(1) == 1);
  const int n = idx.size(0);
  if (n == 0) return;
  hipLaunchKernelGGL(sgmv_kernel, dim3(n), dim3(kBlock), 0,
                     c10::hip::getCurrentHIPStream().stream(),
                     (ushort*)y.data_ptr(), (const ushort*)x.data_ptr(),
                     (const ushort*)A.data_ptr(), (const ushort*)B.data_ptr(),
                     idx.data_ptr<int64_t>(), (float)scale, H, R, out,
                     x.stride(0), y.stride(0));
  HIP_CHECK_KERNEL();
}
