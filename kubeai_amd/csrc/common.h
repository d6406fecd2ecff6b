// Common helpers for kubeai_amd gfx950 (CDNA4) kernels.
//
// Conventions (see /opt/skills/guides/cdna_hip_programming.md):
//  - wave = 64 lanes, hard-coded.
//  - bf16 global loads are vectorized as ushort4/ushort8 (hipcc does not
//    auto-vectorize scalar bf16 loads).
//  - fp32 accumulation everywhere.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <hip/hip_fp8.h>
#include <stdint.h>
#include <c10/hip/HIPStream.h>

#define WAVE_SIZE 64
#define DEV_INLINE __device__ __forceinline__

typedef __hip_bfloat16 bf16_t;

// 16-byte vector of 8 bf16 values (one coalesced lane-load).
typedef ushort __attribute__((ext_vector_type(8))) ushort8;
typedef ushort __attribute__((ext_vector_type(4))) ushort4_t;
typedef float __attribute__((ext_vector_type(4))) float4_t;

DEV_INLINE float bf16_to_f32(ushort u) {
  union {
    uint32_t i;
    float f;
  } c;
  c.i = ((uint32_t)u) << 16;
  return c.f;
}

DEV_INLINE ushort f32_to_bf16(float f) {
  // round-to-nearest-even, matching PyTorch's bf16 cast
  union {
    float f;
    uint32_t i;
  } c;
  c.f = f;
  uint32_t x = c.i;
  uint32_t rb = ((x >> 16) & 1u) + 0x7fffu;
  x += rb;
  return (ushort)(x >> 16);
}

// ---------------- fp8 e5m2 KV-cache element (OCP bf8) ----------------
// e5m2 [s eeeee mm] -> f32 by bit widening: bias 15 -> 127. Matches
// torch.float8_e5m2 exactly incl. subnormals/inf/nan (pure exponent
// remap; subnormals handled by normalizing the 2-bit mantissa).

DEV_INLINE float e5m2_to_f32(unsigned char b) {
  const uint32_t s = ((uint32_t)b & 0x80u) << 24;
  uint32_t e = (b >> 2) & 0x1fu;
  uint32_t m = b & 0x3u;
  uint32_t out;
  if (e == 0) {
    if (m == 0) {
      out = s;  // +-0
    } else {
      // subnormal: value = m * 2^-16; normalize
      int shift = (m & 2u) ? 0 : 1;  // m=2,3 -> msb at bit1; m=1 -> bit0
      e = 127 - 15 + 1 - 1 - shift;  // 2^-15 / 2^-16
      m = (m << (shift + 1)) & 0x3u;  // drop the implicit leading 1
      out = s | (e << 23) | (m << 21);
    }
  } else if (e == 0x1fu) {
    out = s | 0x7f800000u | (m << 21);  // inf / nan
  } else {
    out = s | ((e - 15u + 127u) << 23) | (m << 21);
  }
  union { uint32_t i; float f; } c;
  c.i = out;
  return c.f;
}

// native 2 x f32 vector, the result type of the packed bf8->f32 convert
// (v_cvt_pk_f32_bf8: one instruction turns 2 e5m2 bytes into 2 floats)
typedef float floatx2 __attribute__((ext_vector_type(2)));

// 8 packed e5m2 bytes -> 8 bf16 values (one staging unit).
//
// Route: e5m2 is bit-identical to the TOP BYTE of an IEEE fp16, so
// fp16 = byte<<8 (pure integer move); fp16 -> f32 is one hardware cvt;
// f32 -> bf16 truncates losslessly (every e5m2 value, subnormals and
// inf/nan included, is exactly representable in bf16).
//
// Deliberately NOT via __builtin_amdgcn_cvt_pk_f32_bf8 + bit-level
// truncation: this toolchain mis-simplifies bit manipulation of that
// intrinsic's result (low mantissa bit folded to zero — 129/256 byte
// values corrupt, device-verified in scripts/dbg_cvt.hip; an opaque
// asm barrier does not stop it). Arithmetic (fadd/fma) consumers of
// cvt_pk_f32_bf8 are fine — the decode kernel keeps using it.
DEV_INLINE ushort8 e5m2x8_to_bf16x8(uint64_t raw) {
  ushort8 out;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const ushort h = (ushort)((raw >> (8 * j)) << 8);
    const float f = __half2float(__builtin_bit_cast(__half, h));
    out[j] = (ushort)(__builtin_bit_cast(uint32_t, f) >> 16);
  }
  return out;
}

DEV_INLINE unsigned char f32_to_e5m2(float x) {
  // RNE with saturation to max finite (K/V magnitudes stay far below
  // e5m2's 57344 ceiling in practice)
  return (unsigned char)__hip_cvt_float_to_fp8(x, __HIP_SATFINITE, __HIP_E5M2);
}

// ---------------- wave reductions (64-wide) ----------------

DEV_INLINE float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}

DEV_INLINE float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, 64));
  return v;
}

// Block-level sum reduce across up to 16 waves; `smem` must hold >= 16 floats.
// Every thread returns the total.
DEV_INLINE float block_reduce_sum(float v, float* smem) {
  const int wave = threadIdx.x / WAVE_SIZE;
  const int lane = threadIdx.x % WAVE_SIZE;
  const int n_waves = (blockDim.x + WAVE_SIZE - 1) / WAVE_SIZE;
  v = wave_reduce_sum(v);
  if (lane == 0) smem[wave] = v;
  __syncthreads();
  float total = 0.f;
#pragma unroll
  for (int w = 0; w < 16; ++w)
    if (w < n_waves) total += smem[w];
  return total;
}

// ---------------- hash RNG (kernel/ref-identical spec) ----------------
//
// Sampling uses a counter-based hash RNG so the HIP kernel and the PyTorch
// reference produce bit-identical streams: u64 = splitmix64(key*M ^ idx).
DEV_INLINE uint64_t splitmix64(uint64_t z) {
  z += 0x9e3779b97f4a7c15ull;
  z = (z ^ (z >> 30)) * 0xbf58476d1ce4e5b9ull;
  z = (z ^ (z >> 27)) * 0x94d049bb133111ebull;
  return z ^ (z >> 31);
}

// uniform in (0, 1]
DEV_INLINE float hash_uniform(uint64_t key, uint64_t idx) {
  uint64_t h = splitmix64(key ^ (idx * 0xd1342543de82ef95ull));
  // take top 24 bits -> (0,1] to keep -log well-defined
  uint32_t m = (uint32_t)(h >> 40);
  return ((float)m + 1.0f) * (1.0f / 16777216.0f);
}

#define HIP_CHECK_KERNEL()                                    \
  do {                                                        \
    hipError_t e = hipGetLastError();                         \
    if (e != hipSuccess) {                                    \
      TORCH_CHECK(false, "HIP kernel launch failed: ",        \
                  hipGetErrorString(e));                      \
    }                                                         \
  } while (0)
