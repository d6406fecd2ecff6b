// NeoX rotary embedding, in-place over q and k, one launch.
//
// Semantics: kubeai_amd/ops/ref.py::rope. cos/sin are precomputed on host
// ([max_pos, head_dim] fp32, cos | sin halves) — on-device trig would turn
// this memory-bound op VALU-bound (guide Appendix B).
//
// Grid: one wave per (token, head); lane i handles the rotation pair
// (i, i + hd/2) for hd=128 (one pair per lane), generic loop otherwise.
#include <torch/extension.h>

#include "common.h"

namespace {

__global__ void rope_kernel(
    ushort* __restrict__ q,        // [T, n_q, hd]
    ushort* __restrict__ k,        // [T, n_kv, hd]
    const int32_t* __restrict__ positions,  // [T]
    const float* __restrict__ cos_sin,      // [max_pos, hd]
    const int n_q, const int n_kv, const int hd, const int64_t n_tok,
    const int64_t q_stride, const int64_t k_stride) {
  const int n_heads = n_q + n_kv;
  const int64_t flat = (int64_t)blockIdx.x * (blockDim.x / WAVE_SIZE) +
                       threadIdx.x / WAVE_SIZE;
  const int64_t tok = flat / n_heads;
  if (tok >= n_tok) return;
  const int head = (int)(flat % n_heads);
  const int lane = threadIdx.x % WAVE_SIZE;
  const int half = hd / 2;

  const int pos = positions[tok];
  const float* cs = cos_sin + (int64_t)pos * hd;
  ushort* base = (head < n_q)
                     ? q + tok * q_stride + (int64_t)head * hd
                     : k + tok * k_stride + (int64_t)(head - n_q) * hd;

  for (int i = lane; i < half; i += WAVE_SIZE) {
    const float c = cs[i];
    const float s = cs[half + i];
    const float x1 = bf16_to_f32(base[i]);
    const float x2 = bf16_to_f32(base[half + i]);
    base[i] = f32_to_bf16(x1 * c - x2 * s);
    base[half + i] = f32_to_bf16(x2 * c + x1 * s);
  }
}

}  // namespace

void rope(torch::Tensor q, torch::Tensor k, torch::Tensor positions,
          torch::Tensor cos_sin) {
  // q/k may be row-strided views into the fused qkv projection output
  TORCH_CHECK(q.stride(2) == 1 && q.stride(1) == q.size(2));
  TORCH_CHECK(k.stride(2) == 1 && k.stride(1) == k.size(2));
  TORCH_CHECK(q.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(positions.scalar_type() == torch::kInt32);
  TORCH_CHECK(cos_sin.scalar_type() == torch::kFloat32);
  const int T = q.size(0);
  const int n_q = q.size(1), n_kv = k.size(1), hd = q.size(2);
  TORCH_CHECK(hd % 2 == 0 && cos_sin.size(1) == hd);
  const int64_t total_waves = (int64_t)T * (n_q + n_kv);
  const int waves_per_block = 4;  // 256 threads
  const int64_t blocks = (total_waves + waves_per_block - 1) / waves_per_block;
  // guard tail waves: launch exact multiple; tail handled by tok bound check
  hipLaunchKernelGGL(rope_kernel, dim3((uint32_t)blocks),
                     dim3(waves_per_block * WAVE_SIZE), 0,
                     c10::hip::getCurrentHIPStream().stream(), (ushort*)q.data_ptr(),
                     (ushort*)k.data_ptr(),
                     positions.data_ptr<int32_t>(),
                     cos_sin.data_ptr<float>(), n_q, n_kv, hd, (int64_t)T,
                     q.stride(0), k.stride(0));
  HIP_CHECK_KERNEL();
}
