// RMSNorm kernels for gfx950.
//
// Semantics match kubeai_amd/ops/ref.py::rmsnorm / fused_add_rmsnorm
// (fp32 accumulation, bf16 IO). One workgroup (256 threads) per token row;
// bf16 loads vectorized as ushort8 (16 B/lane); the fused variant does the
// residual-add + norm in a single HBM round trip (reference analog:
// vLLM's fused_add_rms_norm — re-designed here, not ported).
#include <torch/extension.h>

#include "common.h"

namespace {

constexpr int kBlock = 256;
constexpr int kMaxVecPerThread = 8;  // supports H up to 256*8*8 = 16384

template <bool kFused>
__global__ void rmsnorm_kernel(
    ushort* __restrict__ out,        // [T, H] bf16 (== x for fused)
    ushort* __restrict__ x,          // [T, H] bf16
    ushort* __restrict__ residual,   // [T, H] bf16 (fused only; updated)
    const ushort* __restrict__ w,    // [H] bf16
    const float eps, const int H) {
  const int row = blockIdx.x;
  ushort8* xrow = reinterpret_cast<ushort8*>(x + (int64_t)row * H);
  ushort8* rrow = kFused ? reinterpret_cast<ushort8*>(residual + (int64_t)row * H) : nullptr;
  ushort8* orow = reinterpret_cast<ushort8*>(out + (int64_t)row * H);
  const ushort8* wv = reinterpret_cast<const ushort8*>(w);
  const int nvec = H / 8;

  float vals[kMaxVecPerThread][8];
  float ssum = 0.f;
  int n_iter = 0;
  for (int i = threadIdx.x; i < nvec; i += kBlock, ++n_iter) {
    ushort8 xv = xrow[i];
    ushort8 rin;
    if constexpr (kFused) rin = rrow[i];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf16_to_f32(xv[j]);
      if constexpr (kFused) f += bf16_to_f32(rin[j]);
      vals[n_iter][j] = f;
      ssum += f * f;
    }
    if constexpr (kFused) {
      // write residual' = x + residual back
      ushort8 rv;
#pragma unroll
      for (int j = 0; j < 8; ++j) rv[j] = f32_to_bf16(vals[n_iter][j]);
      rrow[i] = rv;
    }
  }

  __shared__ float red[16];
  float total = block_reduce_sum(ssum, red);
  const float inv_rms = rsqrtf(total / (float)H + eps);

  n_iter = 0;
  for (int i = threadIdx.x; i < nvec; i += kBlock, ++n_iter) {
    ushort8 wvv = wv[i];
    ushort8 ov;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      ov[j] = f32_to_bf16(vals[n_iter][j] * inv_rms * bf16_to_f32(wvv[j]));
    orow[i] = ov;
  }
}

}  // namespace

void rmsnorm(torch::Tensor out, torch::Tensor x, torch::Tensor weight,
             double eps) {
  TORCH_CHECK(x.is_contiguous() && out.is_contiguous());
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16);
  const int H = x.size(-1);
  const int T = x.numel() / H;
  TORCH_CHECK(H % 8 == 0 && H <= kBlock * 8 * kMaxVecPerThread);
  hipLaunchKernelGGL((rmsnorm_kernel<false>), dim3(T), dim3(kBlock), 0,
                     c10::hip::getCurrentHIPStream().stream(),
                     (ushort*)out.data_ptr(), (ushort*)x.data_ptr(), nullptr,
                     (const ushort*)weight.data_ptr(), (float)eps, H);
  HIP_CHECK_KERNEL();
}

void fused_add_rmsnorm(torch::Tensor x, torch::Tensor residual,
                       torch::Tensor weight, double eps) {
  TORCH_CHECK(x.is_contiguous() && residual.is_contiguous());
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16);
  const int H = x.size(-1);
  const int T = x.numel() / H;
  TORCH_CHECK(H % 8 == 0 && H <= kBlock * 8 * kMaxVecPerThread);
  hipLaunchKernelGGL((rmsnorm_kernel<true>), dim3(T), dim3(kBlock), 0,
                     c10::hip::getCurrentHIPStream().stream(),
                     (ushort*)x.data_ptr(), (ushort*)x.data_ptr(),
                     (ushort*)residual.data_ptr(),
                     (const ushort*)weight.data_ptr(), (float)eps, H);
  HIP_CHECK_KERNEL();
}
