// Layout probes for the prefill v2 kernel's three hardware assumptions:
//  (a) mfma_f32_32x32x16_bf16 A/B/C fragment mappings (incl. crow)
//  (b) ds_read_b64_tr_b16 over the 4x16-subtiled V layout
//  (c) the in-register P -> PV-A-fragment exchange (cvt_pk + shfl)
//
//   hipcc --offload-arch=gfx950 scripts/dbg_v2.hip -o /tmp/dbg_v2 && /tmp/dbg_v2
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdint>

#define DEV_INLINE __host__ __device__ __forceinline__
typedef unsigned short ushort;
typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) short bf16x4;
typedef __attribute__((ext_vector_type(16))) float f32x16;

DEV_INLINE ushort f32_to_bf16(float f) {
  union { float f; uint32_t i; } c; c.f = f;
  uint32_t x = c.i;
  x += ((x >> 16) & 1u) + 0x7fffu;
  return (ushort)(x >> 16);
}
DEV_INLINE float bf16_to_f32(ushort u) {
  union { uint32_t i; float f; } c; c.i = ((uint32_t)u) << 16;
  return c.f;
}
DEV_INLINE int crow(int reg, int hi) {
  return (reg & 3) + 8 * (reg >> 2) + 4 * hi;
}
__device__ __forceinline__ uint32_t cvt_pk_bf16(float a, float b) {
  uint32_t r;
  asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(a), "v"(b));
  return r;
}

// ---- (a) MFMA fragment layout --------------------------------------
__global__ void probe_mfma(float* out /* [32][32] */,
                           const ushort* A /* [32][16] */,
                           const ushort* B /* [16][32] */) {
  const int lane = threadIdx.x;
  const int hi = lane >> 5, ln31 = lane & 31;
  bf16x8 a, b;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    a[j] = (short)A[ln31 * 16 + hi * 8 + j];        // A[l&31][(l>>5)*8+j]
    b[j] = (short)B[(hi * 8 + j) * 32 + ln31];      // B[(l>>5)*8+j][l&31]
  }
  f32x16 c;
#pragma unroll
  for (int r = 0; r < 16; ++r) c[r] = 0.f;
  c = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 16; ++r) out[crow(r, hi) * 32 + ln31] = c[r];
}

// ---- (b) tr_read over the subtiled V layout ------------------------
// V[64][128] stored as subtiles: elem(row,col) -> (t4*8+h16)*64 + r*16 + c
// then read the PV B-fragment for (ks, nb): lane should get
// V[ks*16 + (l>>5)*8 + jj][nb*32 + (l&31)], jj = 4*tr + j.
__global__ void probe_tr(ushort* out /* [4 ks][4 nb][64 lanes][8 jj] */,
                         const ushort* Vrow /* [64][128] row-major */) {
  __shared__ ushort v_lds[64 * 128];
  const int t = threadIdx.x;  // 64 threads
  for (int u = t; u < 64 * 128 / 8; u += 64) {
    const int tok = u >> 4, c8 = u & 15;
    bf16x8 v = *reinterpret_cast<const bf16x8*>(Vrow + tok * 128 + c8 * 8);
    const int off =
        ((tok >> 2) * 8 + (c8 >> 1)) * 64 + (tok & 3) * 16 + (c8 & 1) * 8;
    *reinterpret_cast<bf16x8*>(&v_lds[off]) = v;
  }
  __syncthreads();
  const int lane = t, hi = lane >> 5;
  typedef __attribute__((address_space(3))) const char as3c;
  // b3-decoded semantics: 4x4 transpose over the 16-lane grid;
  // result[l][j] = lds[addr((l&~15)+((l>>2)&3)+4j) + 2*(l&3)]
  as3c* vbase = (as3c*)(v_lds) +
                (hi * 2048 + ((lane >> 4) & 1) * 128 +
                 ((lane >> 2) & 3) * 32 + (lane & 3) * 8);
  for (int ks = 0; ks < 4; ++ks)
    for (int nb = 0; nb < 4; ++nb) {
      bf16x4 lo4, hi4;
      as3c* a0 = vbase + ks * 4096 + nb * 256;
      asm volatile(
          "ds_read_b64_tr_b16 %0, %2\n\t"
          "ds_read_b64_tr_b16 %1, %2 offset:1024\n\t"
          "s_waitcnt lgkmcnt(0)"
          : "=&v"(lo4), "=&v"(hi4)
          : "v"(a0));
      ushort* o = out + ((ks * 4 + nb) * 64 + lane) * 8;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        o[j] = (ushort)lo4[j];
        o[4 + j] = (ushort)hi4[j];
      }
    }
}

// ---- (b2) decode raw tr_read semantics -----------------------------
// lds16[i] encodes its own index (two passes: low byte, high byte);
// tr_read with a controlled per-lane address reveals exactly which
// element each (lane, j) receives.
__global__ void probe_tr_decode(int* out /* [2 addrmode][64][4] */,
                                int pass /* 0=low,1=high */,
                                int* acc /* accumulate */) {
  __shared__ ushort v_lds[8192];
  const int t = threadIdx.x;
  for (int i = t; i < 8192; i += 64) {
    const int enc = pass == 0 ? (i & 0xff) : (i >> 8);
    v_lds[i] = f32_to_bf16((float)enc);
  }
  __syncthreads();
  typedef __attribute__((address_space(3))) const char as3c;
  for (int mode = 0; mode < 2; ++mode) {
    // mode 0: wave-uniform base 0; mode 1: per-lane base (l>>4)*128 B
    as3c* a0 = (as3c*)(v_lds) + (mode == 1 ? (t >> 4) * 128 : 0);
    bf16x4 r;
    asm volatile("ds_read_b64_tr_b16 %0, %1\n\ts_waitcnt lgkmcnt(0)"
                 : "=v"(r) : "v"(a0));
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int enc = (int)(bf16_to_f32((ushort)r[j]) + 0.5f);
      int* slot = &out[(mode * 64 + t) * 4 + j];
      if (pass == 0) *slot = enc;
      else *slot |= enc << 8;
    }
  }
  (void)acc;
}

// ---- (b3) definitive tr_read decode: unique per-lane addresses -----
// lane l's address = l * 256 bytes; lds[i] = i (self-indexing, split
// low/high passes). Each output value v decodes as:
//   source_lane = v / 128   (whose ADDRESS was used)
//   elem_offset = v % 128   (internal offset applied, in elements)
__global__ void probe_tr_unique(int* out /* [64][4] */, int pass) {
  __shared__ ushort v_lds[8192];
  const int t = threadIdx.x;
  for (int i = t; i < 8192; i += 64) {
    const int enc = pass == 0 ? (i & 0xff) : (i >> 8);
    v_lds[i] = f32_to_bf16((float)enc);
  }
  __syncthreads();
  typedef __attribute__((address_space(3))) const char as3c;
  as3c* a0 = (as3c*)(v_lds) + t * 256;
  bf16x4 r;
  asm volatile("ds_read_b64_tr_b16 %0, %1\n\ts_waitcnt lgkmcnt(0)"
               : "=v"(r) : "v"(a0));
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    const int enc = (int)(bf16_to_f32((ushort)r[j]) + 0.5f);
    int* slot = &out[t * 4 + j];
    if (pass == 0) *slot = enc;
    else *slot |= enc << 8;
  }
}

// ---- (c) P exchange ------------------------------------------------
// st2[sub][reg] encodes P[q][kv] = q*100 + kv (q = l&31 col,
// kv = sub*32 + crow(reg, hi)); after assembly pa[ks] lane l elem jj
// must equal P[l&31][ks*16 + (l>>5)*8 + jj].
__global__ void probe_pex(ushort* out /* [4 ks][64][8] bf16 */) {
  const int lane = threadIdx.x, hi = lane >> 5, ln31 = lane & 31;
  float st2[2][16];
  for (int sub = 0; sub < 2; ++sub)
    for (int r = 0; r < 16; ++r)
      st2[sub][r] = (float)(ln31 * 100 + sub * 32 + crow(r, hi));
  for (int ks = 0; ks < 4; ++ks) {
    const int sub = ks >> 1;
    const int q8_own = (ks & 1) * 2 + hi;
    const uint32_t own0 =
        cvt_pk_bf16(st2[sub][4 * q8_own], st2[sub][4 * q8_own + 1]);
    const uint32_t own1 =
        cvt_pk_bf16(st2[sub][4 * q8_own + 2], st2[sub][4 * q8_own + 3]);
    const int q8_par = (ks & 1) * 2 + 1 - hi;
    const uint32_t snd0 =
        cvt_pk_bf16(st2[sub][4 * q8_par], st2[sub][4 * q8_par + 1]);
    const uint32_t snd1 =
        cvt_pk_bf16(st2[sub][4 * q8_par + 2], st2[sub][4 * q8_par + 3]);
    const uint32_t got0 = __shfl_xor((int)snd0, 32, 64);
    const uint32_t got1 = __shfl_xor((int)snd1, 32, 64);
    uint32_t w[4];
    w[0] = hi ? got0 : own0;
    w[1] = hi ? got1 : own1;
    w[2] = hi ? own0 : got0;
    w[3] = hi ? own1 : got1;
    ushort* o = out + (ks * 64 + lane) * 8;
    for (int i = 0; i < 4; ++i) {
      o[2 * i] = (ushort)(w[i] & 0xffff);
      o[2 * i + 1] = (ushort)(w[i] >> 16);
    }
  }
}

int main() {
  int fails = 0;
  // ---------------- (a)
  {
    ushort *A, *B; float* C;
    hipMallocManaged(&A, 32 * 16 * 2);
    hipMallocManaged(&B, 16 * 32 * 2);
    hipMallocManaged(&C, 32 * 32 * 4);
    float Af[32][16], Bf[16][32];
    for (int m = 0; m < 32; ++m)
      for (int k = 0; k < 16; ++k) {
        Af[m][k] = (float)((m * 16 + k) % 7 - 3);
        A[m * 16 + k] = f32_to_bf16(Af[m][k]);
      }
    for (int k = 0; k < 16; ++k)
      for (int n = 0; n < 32; ++n) {
        Bf[k][n] = (float)((k * 32 + n) % 5 - 2);
        B[k * 32 + n] = f32_to_bf16(Bf[k][n]);
      }
    hipLaunchKernelGGL(probe_mfma, dim3(1), dim3(64), 0, 0, C, A, B);
    hipDeviceSynchronize();
    int bad = 0;
    for (int m = 0; m < 32; ++m)
      for (int n = 0; n < 32; ++n) {
        float want = 0;
        for (int k = 0; k < 16; ++k) want += Af[m][k] * Bf[k][n];
        if (C[m * 32 + n] != want) {
          if (bad < 5)
            printf("(a) C[%d][%d] got %g want %g\n", m, n, C[m * 32 + n], want);
          ++bad;
        }
      }
    printf("(a) mfma layout: %d mismatches\n", bad);
    fails += bad != 0;
  }
  // ---------------- (b)
  {
    ushort *V, *O;
    hipMallocManaged(&V, 64 * 128 * 2);
    hipMallocManaged(&O, 4 * 4 * 64 * 8 * 2);
    for (int r = 0; r < 64; ++r)
      for (int c = 0; c < 128; ++c)
        V[r * 128 + c] = f32_to_bf16((float)((r * 131 + c) % 251));
    hipLaunchKernelGGL(probe_tr, dim3(1), dim3(64), 0, 0, O, V);
    hipDeviceSynchronize();
    int bad = 0;
    for (int ks = 0; ks < 4; ++ks)
      for (int nb = 0; nb < 4; ++nb)
        for (int lane = 0; lane < 64; ++lane)
          for (int jj = 0; jj < 8; ++jj) {
            const int row = ks * 16 + (lane >> 5) * 8 + jj;
            const int col = nb * 32 + (lane & 31);
            ushort want = V[row * 128 + col];
            ushort got = O[(((ks * 4 + nb) * 64) + lane) * 8 + jj];
            if (got != want) {
              if (bad < 8)
                printf("(b) ks=%d nb=%d lane=%d jj=%d got %04x (%g) want "
                       "%04x (%g)\n", ks, nb, lane, jj, got,
                       bf16_to_f32(got), want, bf16_to_f32(want));
              ++bad;
            }
          }
    printf("(b) tr_read layout: %d mismatches of %d\n", bad, 4 * 4 * 64 * 8);
    fails += bad != 0;
  }
  // ---------------- (c)
  {
    ushort* O;
    hipMallocManaged(&O, 4 * 64 * 8 * 2);
    hipLaunchKernelGGL(probe_pex, dim3(1), dim3(64), 0, 0, O);
    hipDeviceSynchronize();
    int bad = 0;
    for (int ks = 0; ks < 4; ++ks)
      for (int lane = 0; lane < 64; ++lane)
        for (int jj = 0; jj < 8; ++jj) {
          const float want =
              (float)((lane & 31) * 100 + ks * 16 + (lane >> 5) * 8 + jj);
          const float got = bf16_to_f32(O[((ks * 64) + lane) * 8 + jj]);
          // bf16 rounding of q*100+kv (< 3200): representable within 16
          const float wb = bf16_to_f32(f32_to_bf16(want));
          if (got != wb) {
            if (bad < 8)
              printf("(c) ks=%d lane=%d jj=%d got %g want %g\n",
                     ks, lane, jj, got, wb);
            ++bad;
          }
        }
    printf("(c) P exchange: %d mismatches of %d\n", bad, 4 * 64 * 8);
    fails += bad != 0;
  }
  // ---------------- (b2) raw tr_read semantics
  {
    int* O;
    hipMallocManaged(&O, 2 * 64 * 4 * 4);
    hipLaunchKernelGGL(probe_tr_decode, dim3(1), dim3(64), 0, 0, O, 0,
                       nullptr);
    hipDeviceSynchronize();
    hipLaunchKernelGGL(probe_tr_decode, dim3(1), dim3(64), 0, 0, O, 1,
                       nullptr);
    hipDeviceSynchronize();
    for (int mode = 0; mode < 2; ++mode) {
      printf("(b2) mode %d (%s):\n", mode,
             mode ? "per-lane base (l>>4)*128B" : "uniform base 0");
      for (int lane : {0, 1, 2, 15, 16, 17, 31, 32, 33, 48, 63}) {
        printf("  lane %2d reads elems:", lane);
        for (int j = 0; j < 4; ++j)
          printf(" %5d", O[(mode * 64 + lane) * 4 + j]);
        printf("\n");
      }
      // candidate formula fits
      int fit1 = 0, fit2 = 0, fit3 = 0;
      for (int l = 0; l < 64; ++l)
        for (int j = 0; j < 4; ++j) {
          const int base = mode ? (l >> 4) * 64 : 0;
          const int got = O[(mode * 64 + l) * 4 + j];
          if (got == base + (l & 15) + j * 16 + (l >> 4) * 64) ++fit1;
          if (got == base + (l & 15) + j * 16) ++fit2;
          if (got == base + (l & 15) * 4 + j) ++fit3;
        }
      printf("  fit[(l&15)+j*16+(l>>4)*64]=%d  fit[(l&15)+j*16]=%d  "
             "fit[(l&15)*4+j]=%d  (of 256)\n", fit1, fit2, fit3);
    }
  }
  // ---------------- (b3) unique-address decode
  {
    int* O;
    hipMallocManaged(&O, 64 * 4 * 4);
    hipLaunchKernelGGL(probe_tr_unique, dim3(1), dim3(64), 0, 0, O, 0);
    hipDeviceSynchronize();
    hipLaunchKernelGGL(probe_tr_unique, dim3(1), dim3(64), 0, 0, O, 1);
    hipDeviceSynchronize();
    printf("(b3) addr_l = l*256B; value -> (src_lane, elem_off):\n");
    for (int l = 0; l < 16; ++l) {
      printf("  lane %2d:", l);
      for (int j = 0; j < 4; ++j) {
        const int v = O[l * 4 + j];
        printf("  (%2d,%3d)", v / 128, v % 128);
      }
      printf("\n");
    }
    printf("  lane 32:");
    for (int j = 0; j < 4; ++j) {
      const int v = O[32 * 4 + j];
      printf("  (%2d,%3d)", v / 128, v % 128);
    }
    printf("\n");
  }
  printf(fails ? "FAIL %d\n" : "ALL OK\n", fails);
  return fails;
}
