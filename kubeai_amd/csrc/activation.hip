// SwiGLU activation: out = silu(gate) * up over packed [T, 2I] input.
// Memory-bound; ushort8 vectorized. Semantics: ops/ref.py::silu_and_mul.
#include <torch/extension.h>

#include "common.h"

namespace {

__global__ void silu_and_mul_kernel(ushort* __restrict__ out,      // [T, I]
                                    const ushort* __restrict__ x,  // [T, 2I]
                                    const int64_t T, const int I) {
  const int nvec = I / 8;
  const int64_t total = T * (int64_t)nvec;
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       idx < total; idx += (int64_t)gridDim.x * blockDim.x) {
    const int64_t row = idx / nvec;
    const int col = (int)(idx % nvec);
    const ushort8* gate =
        reinterpret_cast<const ushort8*>(x + row * 2 * I) + col;
    const ushort8* up =
        reinterpret_cast<const ushort8*>(x + row * 2 * I + I) + col;
    ushort8 g = *gate, u = *up;
    ushort8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gf = bf16_to_f32(g[j]);
      float s = gf / (1.0f + __expf(-gf));
      o[j] = f32_to_bf16(s * bf16_to_f32(u[j]));
    }
    reinterpret_cast<ushort8*>(out + row * I)[col] = o;
  }
}

// GeGLU (gemma family): out = gelu_tanh(gate) * up. Same layout as
// silu_and_mul; semantics ops/ref.py::gelu_and_mul.
__global__ void gelu_and_mul_kernel(ushort* __restrict__ out,
                                    const ushort* __restrict__ x,
                                    const int64_t T, const int I) {
  const int nvec = I / 8;
  const int64_t total = T * (int64_t)nvec;
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       idx < total; idx += (int64_t)gridDim.x * blockDim.x) {
    const int64_t row = idx / nvec;
    const int col = (int)(idx % nvec);
    const ushort8* gate =
        reinterpret_cast<const ushort8*>(x + row * 2 * I) + col;
    const ushort8* up =
        reinterpret_cast<const ushort8*>(x + row * 2 * I + I) + col;
    ushort8 g = *gate, u = *up;
    ushort8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float gf = bf16_to_f32(g[j]);
      // tanh approximation (HF "gelu_pytorch_tanh")
      const float c = 0.7978845608028654f;  // sqrt(2/pi)
      const float t = tanhf(c * (gf + 0.044715f * gf * gf * gf));
      const float act = 0.5f * gf * (1.0f + t);
      o[j] = f32_to_bf16(act * bf16_to_f32(u[j]));
    }
    reinterpret_cast<ushort8*>(out + row * I)[col] = o;
  }
}

}  // namespace

void gelu_and_mul(torch::Tensor out, torch::Tensor x) {
  TORCH_CHECK(x.is_contiguous() && out.is_contiguous());
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16);
  const int64_t T = x.size(0);
  const int I = x.size(1) / 2;
  TORCH_CHECK(I % 8 == 0 && out.size(1) == I);
  if (T == 0) return;
  const int64_t total = T * (I / 8);
  const int block = 256;
  const int grid = (int)std::min<int64_t>((total + block - 1) / block, 2048);
  hipLaunchKernelGGL(gelu_and_mul_kernel, dim3(grid), dim3(block), 0,
                     c10::hip::getCurrentHIPStream().stream(),
                     (ushort*)out.data_ptr(), (const ushort*)x.data_ptr(), T,
                     I);
  HIP_CHECK_KERNEL();
}

void silu_and_mul(torch::Tensor out, torch::Tensor x) {
  TORCH_CHECK(x.is_contiguous() && out.is_contiguous());
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16);
  const int64_t T = x.size(0);
  const int I = x.size(1) / 2;
  TORCH_CHECK(I % 8 == 0 && out.size(1) == I);
  if (T == 0) return;
  const int64_t total = T * (I / 8);
  const int block = 256;
  const int grid = (int)std::min<int64_t>((total + block - 1) / block, 2048);
  hipLaunchKernelGGL(silu_and_mul_kernel, dim3(grid), dim3(block), 0,
                     c10::hip::getCurrentHIPStream().stream(), (ushort*)out.data_ptr(),
                     (const ushort*)x.data_ptr(), T, I);
  HIP_CHECK_KERNEL();
}
