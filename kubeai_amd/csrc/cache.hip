// Paged KV-cache write: scatter new K/V token vectors into cache blocks.
//
// Cache layout [num_blocks, n_kv, block_size, hd]: one (block, kv-head) tile
// is block_size*hd contiguous elements, the coalesced read unit for decode.
// Semantics: kubeai_amd/ops/ref.py::reshape_and_cache.
#include <torch/extension.h>

#include "common.h"

namespace {

__global__ void reshape_and_cache_kernel(
    const ushort* __restrict__ k,   // [T, n_kv, hd]
    const ushort* __restrict__ v,   // [T, n_kv, hd]
    ushort* __restrict__ k_cache,   // [nb, n_kv, bs, hd]
    ushort* __restrict__ v_cache,
    const int64_t* __restrict__ slot_mapping,  // [T]
    const int n_kv, const int bs, const int hd, const int64_t n_tok,
    const int64_t kv_stride) {
  // one wave per (token, kv_head); lane i copies 8 elems (hd=128 -> 2 vec/lane)
  const int64_t flat = (int64_t)blockIdx.x * (blockDim.x / WAVE_SIZE) +
                       threadIdx.x / WAVE_SIZE;
  const int64_t tok = flat / n_kv;
  if (tok >= n_tok) return;
  const int h = (int)(flat % n_kv);
  const int lane = threadIdx.x % WAVE_SIZE;
  const int64_t slot = slot_mapping[tok];
  if (slot < 0) return;
  const int64_t blk = slot / bs;
  const int off = (int)(slot % bs);

  const ushort8* src_k =
      reinterpret_cast<const ushort8*>(k + tok * kv_stride + (int64_t)h * hd);
  const ushort8* src_v =
      reinterpret_cast<const ushort8*>(v + tok * kv_stride + (int64_t)h * hd);
  ushort8* dst_k = reinterpret_cast<ushort8*>(
      k_cache + (((int64_t)blk * n_kv + h) * bs + off) * hd);
  ushort8* dst_v = reinterpret_cast<ushort8*>(
      v_cache + (((int64_t)blk * n_kv + h) * bs + off) * hd);
  const int nvec = hd / 8;
  for (int i = lane; i < nvec; i += WAVE_SIZE) {
    dst_k[i] = src_k[i];
    dst_v[i] = src_v[i];
  }
}

// fp8(e5m2) cache variant: convert bf16 K/V to 1-byte cache elements while
// scattering (pending device validation — kv_cache_dtype=fp8_e5m2 path).
__global__ void reshape_and_cache_fp8_kernel(
    const ushort* __restrict__ k, const ushort* __restrict__ v,
    unsigned char* __restrict__ k_cache, unsigned char* __restrict__ v_cache,
    const int64_t* __restrict__ slot_mapping, const int n_kv, const int bs,
    const int hd, const int64_t n_tok, const int64_t kv_stride) {
  const int64_t flat = (int64_t)blockIdx.x * (blockDim.x / WAVE_SIZE) +
                       threadIdx.x / WAVE_SIZE;
  const int64_t tok = flat / n_kv;
  if (tok >= n_tok) return;
  const int h = (int)(flat % n_kv);
  const int lane = threadIdx.x % WAVE_SIZE;
  const int64_t slot = slot_mapping[tok];
  if (slot < 0) return;
  const int64_t blk = slot / bs;
  const int off = (int)(slot % bs);
  const ushort8* src_k =
      reinterpret_cast<const ushort8*>(k + tok * kv_stride + (int64_t)h * hd);
  const ushort8* src_v =
      reinterpret_cast<const ushort8*>(v + tok * kv_stride + (int64_t)h * hd);
  unsigned char* dst_k = k_cache + (((int64_t)blk * n_kv + h) * bs + off) * hd;
  unsigned char* dst_v = v_cache + (((int64_t)blk * n_kv + h) * bs + off) * hd;
  const int nvec = hd / 8;
  for (int i = lane; i < nvec; i += WAVE_SIZE) {
    const ushort8 kk = src_k[i];
    const ushort8 vv = src_v[i];
    uint64_t ko = 0, vo = 0;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      ko |= (uint64_t)f32_to_e5m2(bf16_to_f32((ushort)kk[j])) << (8 * j);
      vo |= (uint64_t)f32_to_e5m2(bf16_to_f32((ushort)vv[j])) << (8 * j);
    }
    reinterpret_cast<uint64_t*>(dst_k)[i] = ko;
    reinterpret_cast<uint64_t*>(dst_v)[i] = vo;
  }
}

}  // namespace

void reshape_and_cache_fp8(torch::Tensor k, torch::Tensor v,
                           torch::Tensor k_cache, torch::Tensor v_cache,
                           torch::Tensor slot_mapping) {
  TORCH_CHECK(k.stride(2) == 1 && k.stride(1) == k.size(2));
  TORCH_CHECK(k.stride(0) == v.stride(0), "k/v must share row stride");
  TORCH_CHECK(k_cache.is_contiguous() && v_cache.is_contiguous());
  TORCH_CHECK(k_cache.scalar_type() == torch::kFloat8_e5m2);
  const int T = k.size(0), n_kv = k.size(1), hd = k.size(2);
  const int bs = k_cache.size(2);
  TORCH_CHECK(hd % 8 == 0);
  if (T == 0) return;
  const int64_t total_waves = (int64_t)T * n_kv;
  const int waves_per_block = 4;
  const int64_t blocks = (total_waves + waves_per_block - 1) / waves_per_block;
  hipLaunchKernelGGL(reshape_and_cache_fp8_kernel, dim3((uint32_t)blocks),
                     dim3(waves_per_block * WAVE_SIZE), 0,
                     c10::hip::getCurrentHIPStream().stream(),
                     (const ushort*)k.data_ptr(), (const ushort*)v.data_ptr(),
                     (unsigned char*)k_cache.data_ptr(),
                     (unsigned char*)v_cache.data_ptr(),
                     slot_mapping.data_ptr<int64_t>(), n_kv, bs, hd,
                     (int64_t)T, k.stride(0));
  HIP_CHECK_KERNEL();
}

void reshape_and_cache(torch::Tensor k, torch::Tensor v, torch::Tensor k_cache,
                       torch::Tensor v_cache, torch::Tensor slot_mapping) {
  TORCH_CHECK(k.stride(2) == 1 && k.stride(1) == k.size(2));
  TORCH_CHECK(k.stride(0) == v.stride(0), "k/v must share row stride");
  TORCH_CHECK(v.stride(2) == 1 && v.stride(1) == v.size(2));
  TORCH_CHECK(k_cache.is_contiguous() && v_cache.is_contiguous());
  TORCH_CHECK(slot_mapping.scalar_type() == torch::kInt64);
  const int T = k.size(0), n_kv = k.size(1), hd = k.size(2);
  const int bs = k_cache.size(2);
  TORCH_CHECK(hd % 8 == 0);
  if (T == 0) return;
  const int64_t total_waves = (int64_t)T * n_kv;
  const int waves_per_block = 4;
  const int64_t blocks = (total_waves + waves_per_block - 1) / waves_per_block;
  // tail guard: slot_mapping index must stay in range
  hipLaunchKernelGGL(reshape_and_cache_kernel, dim3((uint32_t)blocks),
                     dim3(waves_per_block * WAVE_SIZE), 0,
                     c10::hip::getCurrentHIPStream().stream(),
                     (const ushort*)k.data_ptr(), (const ushort*)v.data_ptr(),
                     (ushort*)k_cache.data_ptr(), (ushort*)v_cache.data_ptr(),
                     slot_mapping.data_ptr<int64_t>(), n_kv, bs, hd,
                     (int64_t)T, k.stride(0));
  HIP_CHECK_KERNEL();
}
