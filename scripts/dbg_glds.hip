// Probe: global_load_lds_dwordx4 semantics on gfx950.
// Question: with a uniform LDS base operand, where does lane i's 16 B land?
// Hypothesis: LDS addr = M0(base) + instr_offset + lane*16 (contiguous).
// Also checks two sequential DMAs with different LDS bases (compiler must
// re-set M0) and vmcnt-based completion of a double-buffer pattern.
//
// Build+run:
//   hipcc --offload-arch=gfx950 -O3 scripts/dbg_glds.hip -o /tmp/dbg_glds
//   /tmp/dbg_glds
#include <hip/hip_runtime.h>

#include <cstdio>
#include <vector>

#define CHK(x)                                                    \
  do {                                                            \
    hipError_t e = (x);                                           \
    if (e != hipSuccess) {                                        \
      printf("HIP error %s at %d\n", hipGetErrorString(e), __LINE__); \
      return 1;                                                   \
    }                                                             \
  } while (0)

typedef __attribute__((address_space(1))) const void as1_void;
typedef __attribute__((address_space(3))) void as3_void;

__global__ void probe_kernel(const float* __restrict__ g, float* out,
                             int* order_ok) {
  __shared__ float lds_a[256];
  __shared__ float lds_b[256];
  const int lane = threadIdx.x;
  // each lane sources its own 16 B from g at lane*16B (4 floats)
  const float* src_a = g + lane * 4;
  const float* src_b = g + 1024 + lane * 4;
  __builtin_amdgcn_global_load_lds((as1_void*)src_a, (as3_void*)lds_a, 16, 0,
                                   0);
  __builtin_amdgcn_global_load_lds((as1_void*)src_b, (as3_void*)lds_b, 16, 0,
                                   0);
  asm volatile("s_waitcnt vmcnt(0)");
  __builtin_amdgcn_s_barrier();
  // dump both buffers: out[0..255] = lds_a, out[256..511] = lds_b
  for (int i = lane; i < 256; i += 64) {
    out[i] = lds_a[i];
    out[256 + i] = lds_b[i];
  }
  if (lane == 0) *order_ok = 1;
}

// double-buffer pattern: DMA tile n+1 while "computing" (summing) tile n,
// wait vmcnt(2) style partial waits. 4 tiles of 64 lanes * 16 B.
__global__ void dbuf_kernel(const float* __restrict__ g, float* out) {
  __shared__ float buf[2][256];
  const int lane = threadIdx.x;
  __builtin_amdgcn_global_load_lds((as1_void*)(g + lane * 4),
                                   (as3_void*)buf[0], 16, 0, 0);
  float acc = 0.f;
  for (int t = 0; t < 4; ++t) {
    if (t + 1 < 4)
      __builtin_amdgcn_global_load_lds(
          (as1_void*)(g + (t + 1) * 256 + lane * 4),
          (as3_void*)buf[(t + 1) & 1], 16, 0, 0);
    // wait for all but the just-issued DMA (1 outstanding allowed)
    if (t + 1 < 4)
      asm volatile("s_waitcnt vmcnt(1)");
    else
      asm volatile("s_waitcnt vmcnt(0)");
    for (int i = 0; i < 4; ++i) acc += buf[t & 1][lane * 4 + i];
  }
  out[lane] = acc;
}

int main() {
  float* g;
  float* out;
  int* ok;
  CHK(hipMalloc(&g, 8192 * 4));
  CHK(hipMalloc(&out, 512 * 4));
  CHK(hipMalloc(&ok, 4));
  std::vector<float> h(8192);
  for (int i = 0; i < 8192; ++i) h[i] = (float)i;
  CHK(hipMemcpy(g, h.data(), 8192 * 4, hipMemcpyHostToDevice));
  probe_kernel<<<1, 64>>>(g, out, ok);
  CHK(hipDeviceSynchronize());
  std::vector<float> o(512);
  CHK(hipMemcpy(o.data(), out, 512 * 4, hipMemcpyDeviceToHost));
  int bad = 0;
  for (int i = 0; i < 256; ++i) {
    if (o[i] != (float)i) {
      if (bad < 4) printf("A[%d]=%g want %d\n", i, o[i], i);
      ++bad;
    }
    if (o[256 + i] != (float)(1024 + i)) {
      if (bad < 4) printf("B[%d]=%g want %d\n", i, o[256 + i], 1024 + i);
      ++bad;
    }
  }
  printf("probe contiguous-lane placement: %s (%d bad)\n",
         bad ? "FAIL" : "OK", bad);

  dbuf_kernel<<<1, 64>>>(g, out);
  CHK(hipDeviceSynchronize());
  CHK(hipMemcpy(o.data(), out, 64 * 4, hipMemcpyDeviceToHost));
  int bad2 = 0;
  for (int l = 0; l < 64; ++l) {
    float want = 0.f;
    for (int t = 0; t < 4; ++t)
      for (int i = 0; i < 4; ++i) want += (float)(t * 256 + l * 4 + i);
    if (o[l] != want) ++bad2;
  }
  printf("double-buffer vmcnt pattern: %s (%d bad)\n", bad2 ? "FAIL" : "OK",
         bad2);
  return bad + bad2 ? 1 : 0;
}
