// Fused fp8 (OCP e4m3) quantization kernels.
//
// The W8A8 path's cost on MI355X is NOT the GEMM — it is the per-linear
// dynamic activation quantization (amax + scale + cast = extra kernel
// launches and HBM round trips). These kernels emit fp8 + per-token scales
// directly from the PRODUCING op, so the quantization is free:
//   rmsnorm_fp8 / fused_add_rmsnorm_fp8  -> feeds qkv_proj / gate_up_proj
//   silu_and_mul_fp8                     -> feeds down_proj
//   quant_fp8 (one pass)                 -> feeds o_proj (attention out)
// Scale convention: per row, scale = max(|x|)/448; y = x/scale, RNE cast
// (matches torch .to(float8_e4m3fn) + torch._scaled_mm rowwise scale_a).
#include <hip/hip_fp8.h>
#include <torch/extension.h>

#include "common.h"

namespace {

constexpr int kBlock = 256;
constexpr int kMaxVecPerThread = 16;  // rows up to 32768 elems (70B silu: I=28672)

DEV_INLINE unsigned char f32_to_fp8(float x) {
  return (unsigned char)__hip_cvt_float_to_fp8(x, __HIP_SATFINITE, __HIP_E4M3);
}

// block-wide amax over per-thread partials (smem >= 16 floats)
DEV_INLINE float block_amax(float v, float* smem) {
  const int wave = threadIdx.x / WAVE_SIZE;
  const int lane = threadIdx.x % WAVE_SIZE;
  const int n_waves = (blockDim.x + WAVE_SIZE - 1) / WAVE_SIZE;
  v = wave_reduce_max(v);
  if (lane == 0) smem[wave] = v;
  __syncthreads();
  float m = 0.f;
#pragma unroll
  for (int w = 0; w < 16; ++w)
    if (w < n_waves) m = fmaxf(m, smem[w]);
  return m;
}

// ---- rmsnorm -> fp8 (optionally fused residual add) ----
template <bool kFused>
__global__ void rmsnorm_fp8_kernel(
    unsigned char* __restrict__ out,  // [T, H] fp8
    float* __restrict__ out_scale,    // [T]
    ushort* __restrict__ x,           // [T, H] bf16
    ushort* __restrict__ residual,    // [T, H] bf16 (fused; updated)
    const ushort* __restrict__ w,     // [H]
    const float eps, const int H) {
  const int row = blockIdx.x;
  ushort8* xrow = reinterpret_cast<ushort8*>(x + (int64_t)row * H);
  ushort8* rrow =
      kFused ? reinterpret_cast<ushort8*>(residual + (int64_t)row * H) : nullptr;
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
      ushort8 rv;
#pragma unroll
      for (int j = 0; j < 8; ++j) rv[j] = f32_to_bf16(vals[n_iter][j]);
      rrow[i] = rv;
    }
  }

  __shared__ float red[16];
  const float total = block_reduce_sum(ssum, red);
  const float inv_rms = rsqrtf(total / (float)H + eps);

  // normalize in registers, track amax
  float amax = 0.f;
  n_iter = 0;
  for (int i = threadIdx.x; i < nvec; i += kBlock, ++n_iter) {
    ushort8 wvv = wv[i];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      vals[n_iter][j] *= inv_rms * bf16_to_f32(wvv[j]);
      amax = fmaxf(amax, fabsf(vals[n_iter][j]));
    }
  }
  __syncthreads();  // red[] reuse
  const float gmax = fmaxf(block_amax(amax, red), 1e-6f);
  const float scale = gmax / 448.0f;
  if (threadIdx.x == 0) out_scale[row] = scale;
  const float inv_scale = 448.0f / gmax;

  n_iter = 0;
  for (int i = threadIdx.x; i < nvec; i += kBlock, ++n_iter) {
    uchar2 packed[4];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      packed[j].x = f32_to_fp8(vals[n_iter][2 * j] * inv_scale);
      packed[j].y = f32_to_fp8(vals[n_iter][2 * j + 1] * inv_scale);
    }
    *reinterpret_cast<uint2*>(out + (int64_t)row * H + i * 8) =
        *reinterpret_cast<uint2*>(packed);
  }
}

// ---- silu(gate) * up -> fp8 ----
__global__ void silu_and_mul_fp8_kernel(
    unsigned char* __restrict__ out,  // [T, I]
    float* __restrict__ out_scale,    // [T]
    const ushort* __restrict__ xin,   // [T, 2I]
    const int I) {
  const int row = blockIdx.x;
  const ushort8* gate = reinterpret_cast<const ushort8*>(xin + (int64_t)row * 2 * I);
  const ushort8* up =
      reinterpret_cast<const ushort8*>(xin + (int64_t)row * 2 * I + I);
  const int nvec = I / 8;

  float vals[kMaxVecPerThread][8];
  float amax = 0.f;
  int n_iter = 0;
  for (int i = threadIdx.x; i < nvec; i += kBlock, ++n_iter) {
    ushort8 g = gate[i], u = up[i];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float gf = bf16_to_f32(g[j]);
      const float s = gf / (1.0f + __expf(-gf));
      const float v = s * bf16_to_f32(u[j]);
      vals[n_iter][j] = v;
      amax = fmaxf(amax, fabsf(v));
    }
  }
  __shared__ float red[16];
  const float gmax = fmaxf(block_amax(amax, red), 1e-6f);
  const float scale = gmax / 448.0f;
  if (threadIdx.x == 0) out_scale[row] = scale;
  const float inv_scale = 448.0f / gmax;
  n_iter = 0;
  for (int i = threadIdx.x; i < nvec; i += kBlock, ++n_iter) {
    uchar2 packed[4];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      packed[j].x = f32_to_fp8(vals[n_iter][2 * j] * inv_scale);
      packed[j].y = f32_to_fp8(vals[n_iter][2 * j + 1] * inv_scale);
    }
    *reinterpret_cast<uint2*>(out + (int64_t)row * I + i * 8) =
        *reinterpret_cast<uint2*>(packed);
  }
}

// ---- plain bf16 -> fp8 row quant (attention output) ----
__global__ void quant_fp8_kernel(unsigned char* __restrict__ out,
                                 float* __restrict__ out_scale,
                                 const ushort* __restrict__ xin, const int H) {
  const int row = blockIdx.x;
  const ushort8* xrow = reinterpret_cast<const ushort8*>(xin + (int64_t)row * H);
  const int nvec = H / 8;
  float vals[kMaxVecPerThread][8];
  float amax = 0.f;
  int n_iter = 0;
  for (int i = threadIdx.x; i < nvec; i += kBlock, ++n_iter) {
    ushort8 xv = xrow[i];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float f = bf16_to_f32(xv[j]);
      vals[n_iter][j] = f;
      amax = fmaxf(amax, fabsf(f));
    }
  }
  __shared__ float red[16];
  const float gmax = fmaxf(block_amax(amax, red), 1e-6f);
  const float scale = gmax / 448.0f;
  if (threadIdx.x == 0) out_scale[row] = scale;
  const float inv_scale = 448.0f / gmax;
  n_iter = 0;
  for (int i = threadIdx.x; i < nvec; i += kBlock, ++n_iter) {
    uchar2 packed[4];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      packed[j].x = f32_to_fp8(vals[n_iter][2 * j] * inv_scale);
      packed[j].y = f32_to_fp8(vals[n_iter][2 * j + 1] * inv_scale);
    }
    *reinterpret_cast<uint2*>(out + (int64_t)row * H + i * 8) =
        *reinterpret_cast<uint2*>(packed);
  }
}

void check_row_op(const torch::Tensor& x, int max_h) {
  TORCH_CHECK(x.is_contiguous());
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(x.size(-1) % 8 == 0 && x.size(-1) <= max_h);
}

}  // namespace

void rmsnorm_fp8(torch::Tensor out, torch::Tensor out_scale, torch::Tensor x,
                 torch::Tensor weight, double eps) {
  check_row_op(x, kBlock * 8 * kMaxVecPerThread);
  const int H = x.size(-1);
  const int T = x.numel() / H;
  hipLaunchKernelGGL((rmsnorm_fp8_kernel<false>), dim3(T), dim3(kBlock), 0,
                     c10::hip::getCurrentHIPStream().stream(),
                     (unsigned char*)out.data_ptr(),
                     out_scale.data_ptr<float>(), (ushort*)x.data_ptr(),
                     nullptr, (const ushort*)weight.data_ptr(), (float)eps, H);
  HIP_CHECK_KERNEL();
}

void fused_add_rmsnorm_fp8(torch::Tensor out, torch::Tensor out_scale,
                           torch::Tensor x, torch::Tensor residual,
                           torch::Tensor weight, double eps) {
  check_row_op(x, kBlock * 8 * kMaxVecPerThread);
  TORCH_CHECK(residual.is_contiguous());
  const int H = x.size(-1);
  const int T = x.numel() / H;
  hipLaunchKernelGGL((rmsnorm_fp8_kernel<true>), dim3(T), dim3(kBlock), 0,
                     c10::hip::getCurrentHIPStream().stream(),
                     (unsigned char*)out.data_ptr(),
                     out_scale.data_ptr<float>(), (ushort*)x.data_ptr(),
                     (ushort*)residual.data_ptr(),
                     (const ushort*)weight.data_ptr(), (float)eps, H);
  HIP_CHECK_KERNEL();
}

void silu_and_mul_fp8(torch::Tensor out, torch::Tensor out_scale,
                      torch::Tensor x) {
  check_row_op(x, 2 * kBlock * 8 * kMaxVecPerThread);
  const int I = x.size(1) / 2;
  TORCH_CHECK(I % 8 == 0 && I <= kBlock * 8 * kMaxVecPerThread);
  const int T = x.size(0);
  if (T == 0) return;
  hipLaunchKernelGGL(silu_and_mul_fp8_kernel, dim3(T), dim3(kBlock), 0,
                     c10::hip::getCurrentHIPStream().stream(),
                     (unsigned char*)out.data_ptr(),
                     out_scale.data_ptr<float>(),
                     (const ushort*)x.data_ptr(), I);
  HIP_CHECK_KERNEL();
}

void quant_fp8(torch::Tensor out, torch::Tensor out_scale, torch::Tensor x) {
  check_row_op(x, kBlock * 8 * kMaxVecPerThread);
  const int H = x.size(-1);
  const int T = x.numel() / H;
  if (T == 0) return;
  hipLaunchKernelGGL(quant_fp8_kernel, dim3(T), dim3(kBlock), 0,
                     c10::hip::getCurrentHIPStream().stream(),
                     (unsigned char*)out.data_ptr(),
                     out_scale.data_ptr<float>(),
                     (const ushort*)x.data_ptr(), H);
  HIP_CHECK_KERNEL();
}
