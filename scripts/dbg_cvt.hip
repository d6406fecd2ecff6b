// Standalone check of e5m2x8_to_bf16x8 (packed bf8->f32 convert + exact
// truncation) against the scalar reference path (e5m2_to_f32 + RNE
// f32_to_bf16) for all 256 byte values, plus raw cvt_pk byte-order probe.
//
//   hipcc --offload-arch=gfx950 scripts/dbg_cvt.hip -o /tmp/dbg_cvt && /tmp/dbg_cvt
#include <hip/hip_runtime.h>
#include <hip/hip_fp16.h>
#include <cstdio>
#include <cstdint>

#define DEV_INLINE __device__ __forceinline__
typedef unsigned short ushort;
typedef ushort ushort8 __attribute__((ext_vector_type(8)));
typedef float floatx2 __attribute__((ext_vector_type(2)));

DEV_INLINE ushort f32_to_bf16(float f) {
  union { float f; uint32_t i; } c; c.f = f;
  uint32_t x = c.i;
  uint32_t rb = ((x >> 16) & 1u) + 0x7fffu;
  x += rb;
  return (ushort)(x >> 16);
}

DEV_INLINE float e5m2_to_f32(unsigned char b) {
  const uint32_t s = ((uint32_t)b & 0x80u) << 24;
  uint32_t e = (b >> 2) & 0x1fu;
  uint32_t m = b & 0x3u;
  uint32_t out;
  if (e == 0) {
    if (m == 0) { out = s; }
    else {
      int shift = (m & 2u) ? 0 : 1;
      e = 127 - 15 + 1 - 1 - shift;
      m = (m << (shift + 1)) & 0x3u;
      out = s | (e << 23) | (m << 21);
    }
  } else if (e == 0x1fu) {
    out = s | 0x7f800000u | (m << 21);
  } else {
    out = s | ((e - 15 + 127) << 23) | (m << 21);
  }
  union { uint32_t i; float f; } c; c.i = out;
  return c.f;
}

DEV_INLINE ushort8 e5m2x8_new(uint64_t raw) {
  // production route (common.h): byte<<8 == the same value as fp16,
  // fp16->f32 hardware cvt, exact truncation to bf16
  ushort8 out;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const ushort h = (ushort)((raw >> (8 * j)) << 8);
    const float f = __half2float(__builtin_bit_cast(__half, h));
    out[j] = (ushort)(__builtin_bit_cast(uint32_t, f) >> 16);
  }
  return out;
}

DEV_INLINE ushort8 e5m2x8_cvt_trunc(uint64_t raw) {
  // KNOWN-BAD on this toolchain (kept to document the bug): bit-level
  // truncation of __builtin_amdgcn_cvt_pk_f32_bf8 results loses the low
  // mantissa bit (129/256 byte values corrupt); an asm barrier between
  // the cvt and the truncation does NOT prevent it. Arithmetic
  // consumers of the cvt are unaffected (decode kernel, cos=1.0).
  ushort8 out;
  const uint32_t w[2] = {(uint32_t)raw, (uint32_t)(raw >> 32)};
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    floatx2 lo = __builtin_amdgcn_cvt_pk_f32_bf8(w[i], false);
    floatx2 hi = __builtin_amdgcn_cvt_pk_f32_bf8(w[i], true);
    asm("" : "+v"(lo), "+v"(hi));
    out[4 * i + 0] = (ushort)(__builtin_bit_cast(uint32_t, lo.x) >> 16);
    out[4 * i + 1] = (ushort)(__builtin_bit_cast(uint32_t, lo.y) >> 16);
    out[4 * i + 2] = (ushort)(__builtin_bit_cast(uint32_t, hi.x) >> 16);
    out[4 * i + 3] = (ushort)(__builtin_bit_cast(uint32_t, hi.y) >> 16);
  }
  return out;
}

__global__ void k(ushort* got, ushort* want, float* raw_f32, int* bad_cvt) {
  // 256 bytes -> 32 uint64 staging units
  for (int u = 0; u < 32; ++u) {
    uint64_t raw = 0;
    for (int j = 0; j < 8; ++j)
      raw |= (uint64_t)(unsigned char)(u * 8 + j) << (8 * j);
    ushort8 g = e5m2x8_new(raw);
    ushort8 gc = e5m2x8_cvt_trunc(raw);
    for (int j = 0; j < 8; ++j) {
      got[u * 8 + j] = g[j];
      want[u * 8 + j] = f32_to_bf16(e5m2_to_f32((unsigned char)(u * 8 + j)));
      if (gc[j] != want[u * 8 + j]) atomicAdd(bad_cvt, 1);
    }
  }
  // raw byte-order probe: bytes 0x3c,0x40,0x44,0x48 = 1.0,2.0,4.0,8.0
  uint32_t w = 0x48444038u;  // little-endian bytes: 0x38,0x40,0x44,0x48
  floatx2 lo = __builtin_amdgcn_cvt_pk_f32_bf8(w, false);
  floatx2 hi = __builtin_amdgcn_cvt_pk_f32_bf8(w, true);
  raw_f32[0] = lo.x; raw_f32[1] = lo.y; raw_f32[2] = hi.x; raw_f32[3] = hi.y;
}

int main() {
  ushort *got, *want; float* rf; int* bad_cvt;
  hipMallocManaged(&got, 256 * 2);
  hipMallocManaged(&want, 256 * 2);
  hipMallocManaged(&rf, 4 * 4);
  hipMallocManaged(&bad_cvt, 4);
  *bad_cvt = 0;
  hipLaunchKernelGGL(k, dim3(1), dim3(1), 0, 0, got, want, rf, bad_cvt);
  hipDeviceSynchronize();
  int bad = 0;
  for (int i = 0; i < 256; ++i) {
    if (got[i] != want[i]) {
      if (bad < 16)
        printf("byte %02x: got %04x want %04x\n", i, got[i], want[i]);
      ++bad;
    }
  }
  printf("mismatches: %d / 256\n", bad);
  printf("byte order probe (expect 0.5 2 4 8): %g %g %g %g\n",
         rf[0], rf[1], rf[2], rf[3]);
  printf("known-bad cvt_pk+trunc mismatches (documents toolchain bug): %d\n",
         *bad_cvt);
  return bad != 0;
}
