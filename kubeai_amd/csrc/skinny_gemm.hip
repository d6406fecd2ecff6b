// Skinny-M streaming GEMM: y[M,N] = x[M,K] @ W[N,K]^T for decode batches
// (M <= 64), bf16 in / bf16 out, fp32 accumulation.
//
// Why hand-written: at M <= 64 these GEMMs are pure weight streams
// (arithmetic intensity = M flops/byte << machine balance), yet hipBLASLt
// sits at 35-62% of HBM on the qkv/o/down shapes (TunableOp-tuned; see
// profiles/r01_results.md). This kernel streams W exactly once at near-HBM
// rate and keeps x in L2.
//
// Structure (MI355X-first):
//  - workgroup = (n_tile of 64 W rows, k_split); 4 waves; wave w owns the
//    16-column n-subtile w. Output tile 64(M) x 64(N).
//  - MFMA v_mfma_f32_16x16x32_bf16; A/B fragments loaded DIRECTLY from
//    global memory: W rows are consumed densely along k, so the per-lane
//    16 B fragment gathers coalesce in L2/HBM lines; x (<=32 KB) lives in
//    L2 after the first touch. No LDS, no barriers, ~60 VGPRs -> high
//    occupancy to cover the stream latency.
//  - K is split across gridDim.y workgroups (>= 1024 in flight); fp32
//    partials land in the caller's workspace via global atomics, finalized
//    to bf16 by skinny_gemm_finalize (ksplit == 1 stores bf16 directly).
#include <torch/extension.h>

#include "common.h"

namespace {

constexpr int kBN = 64;   // W rows (output cols) per workgroup
constexpr int kBK = 64;   // k chunk per iteration (2 MFMA k-steps)

typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

template <int MT>  // number of 16-row m-subtiles (M_pad = MT*16)
__launch_bounds__(256) __global__ void skinny_gemm_kernel(
    float* __restrict__ y_f32,        // [M, N] accumulator (ksplit > 1)
    ushort* __restrict__ y_bf16,      // [M, N] direct output (ksplit == 1)
    const ushort* __restrict__ x,     // [M, K]
    const ushort* __restrict__ w,     // [N, K]
    const int M, const int N, const int K, const int k_split) {
  const int n_tile = blockIdx.x;
  const int ks = blockIdx.y;
  const int wave = threadIdx.x / WAVE_SIZE;
  const int lane = threadIdx.x % WAVE_SIZE;
  const int n0 = n_tile * kBN + wave * 16;  // this wave's 16 output cols

  const int k_chunk_total = K / kBK;
  const int per_split = (k_chunk_total + k_split - 1) / k_split;
  const int kc_lo = ks * per_split;
  const int kc_hi = min(k_chunk_total, kc_lo + per_split);

  // Double-buffered XOR-swizzled tiles, filled with async
  // global_load_lds (16 B, wave-uniform LDS dest + per-lane PRE-SWIZZLED
  // source address — guide m173): one barrier per k-chunk, loads for
  // chunk kc+1 fly while chunk kc computes.
  __shared__ ushort w_lds[2][kBN * kBK];
  __shared__ ushort x_lds[2][64 * kBK];

  f32x4 acc[MT];
#pragma unroll
  for (int mt = 0; mt < MT; ++mt) acc[mt] = {0.f, 0.f, 0.f, 0.f};

  const int kfrag16 = (lane >> 4) * 16;  // byte offset of the lane's k range

  // staging geometry: 512 16B vectors per tile; this thread covers vector
  // ids {tid, tid+256}; source byte offset is swizzled so the linear LDS
  // write lands in swizzled layout
  const int v_lo = threadIdx.x;
  auto stage = [&](int buf, int kc) {
#pragma unroll
    for (int h = 0; h < 2; ++h) {
      const int v = v_lo + h * 256;
      const int row = v >> 3;
      const int col8 = v & 7;
      const int src_off = (col8 * 16) ^ ((row & 7) << 4);  // within 128B
      const char* wsrc = reinterpret_cast<const char*>(
          w + (int64_t)(n_tile * kBN + row) * K + kc * kBK) + src_off;
      const int xrow = min(row, M - 1);
      const char* xsrc = reinterpret_cast<const char*>(
          x + (int64_t)xrow * K + kc * kBK) + src_off;
      // LDS dest is wave-uniform: HW adds lane*16; this wave's 64
      // vectors start at vector id (h*256 + wave*64)
      const int wave_v0 = h * 256 + wave * 64;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)wsrc,
          (__attribute__((address_space(3))) uint32_t*)&w_lds[buf][wave_v0 * 8],
          16, 0, 0);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)xsrc,
          (__attribute__((address_space(3))) uint32_t*)&x_lds[buf][wave_v0 * 8],
          16, 0, 0);
    }
  };

  int cur = 0;
  stage(0, kc_lo);
  __syncthreads();  // compiler drains vmcnt before the barrier
  for (int kc = kc_lo; kc < kc_hi; ++kc) {
    if (kc + 1 < kc_hi) stage(cur ^ 1, kc + 1);
#pragma unroll
    for (int half = 0; half < 2; ++half) {  // two k-steps of 32
      const int kb = half * 64;  // byte offset of this k-step in the row
      const int wrow = wave * 16 + (lane & 15);
      const bf16x8 b_frag = *reinterpret_cast<const bf16x8*>(
          reinterpret_cast<const char*>(w_lds[cur]) +
          ((wrow * kBK * 2 + kb + kfrag16) ^ ((wrow & 7) << 4)));
#pragma unroll
      for (int mt = 0; mt < MT; ++mt) {
        const int arow = mt * 16 + (lane & 15);
        const bf16x8 a_frag = *reinterpret_cast<const bf16x8*>(
            reinterpret_cast<const char*>(x_lds[cur]) +
            ((arow * kBK * 2 + kb + kfrag16) ^ ((arow & 7) << 4)));
        acc[mt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag, b_frag,
                                                          acc[mt], 0, 0, 0);
      }
    }
    __syncthreads();
    cur ^= 1;
  }

  // C layout: lane holds rows (lane>>4)*4+i, col lane&15 of each 16x16 tile
#pragma unroll
  for (int mt = 0; mt < MT; ++mt)
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int row = mt * 16 + (lane >> 4) * 4 + i;
      if (row >= M) continue;
      const int col = n0 + (lane & 15);
      if (k_split == 1) {
        y_bf16[(int64_t)row * N + col] = f32_to_bf16(acc[mt][i]);
      } else {
        atomicAdd(&y_f32[(int64_t)row * N + col], acc[mt][i]);
      }
    }
}

__global__ void skinny_gemm_finalize(ushort* __restrict__ y_bf16,
                                     const float* __restrict__ y_f32,
                                     const int64_t total) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x)
    y_bf16[i] = f32_to_bf16(y_f32[i]);
}

}  // namespace

void skinny_gemm(torch::Tensor y, torch::Tensor x, torch::Tensor w) {
  TORCH_CHECK(x.is_contiguous() && w.is_contiguous() && y.is_contiguous());
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(w.scalar_type() == torch::kBFloat16);
  const int M = x.size(0), K = x.size(1), N = w.size(0);
  TORCH_CHECK(w.size(1) == K && y.size(0) == M && y.size(1) == N);
  TORCH_CHECK(M >= 1 && M <= 64, "skinny_gemm supports M in [1, 64]");
  TORCH_CHECK(K % kBK == 0 && N % kBN == 0);
  const int MT = (M + 15) / 16;
  const int n_tiles = N / kBN;
  // enough workgroups to fill 256 CUs
  int k_split = 1;
  if (n_tiles < 1024)
    k_split = std::min((1024 + n_tiles - 1) / n_tiles, K / kBK);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  torch::Tensor y_f32;
  float* y_f32_ptr = nullptr;
  if (k_split > 1) {
    y_f32 = torch::zeros(
        {M, N},
        torch::TensorOptions().device(x.device()).dtype(torch::kFloat32));
    y_f32_ptr = y_f32.data_ptr<float>();
  }
  dim3 grid(n_tiles, k_split), block(256);
#define LAUNCH(MM)                                                       \
  hipLaunchKernelGGL((skinny_gemm_kernel<MM>), grid, block, 0, stream,   \
                     y_f32_ptr, (ushort*)y.data_ptr(),                   \
                     (const ushort*)x.data_ptr(),                        \
                     (const ushort*)w.data_ptr(), M, N, K, k_split)
  switch (MT) {
    case 1: LAUNCH(1); break;
    case 2: LAUNCH(2); break;
    case 3: LAUNCH(3); break;
    case 4: LAUNCH(4); break;
  }
#undef LAUNCH
  HIP_CHECK_KERNEL();
  if (k_split > 1) {
    const int64_t total = (int64_t)M * N;
    hipLaunchKernelGGL(skinny_gemm_finalize,
                       dim3((uint32_t)std::min<int64_t>((total + 255) / 256, 2048)),
                       dim3(256), 0, stream, (ushort*)y.data_ptr(), y_f32_ptr,
                       total);
    HIP_CHECK_KERNEL();
  }
}
