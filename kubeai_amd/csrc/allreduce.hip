// One-shot fused all-reduce + residual-add + RMSNorm over xGMI (gfx950).
//
// SURVEY.md hard part #2: at TP=8 the decode step is dominated by the
// latency of many small (T x H) all-reduces. A ring pays ~2*(N-1)/N of
// the tensor over ONE 153 GB/s xGMI link plus multiple kernel rounds;
// for small T the bound is LATENCY, not bandwidth. This is the one-shot
// alternative (vLLM-custom-allreduce shape, re-derived for CDNA4):
//
//   k1: every rank copies its partial tensor into its own IPC-shared
//       buffer and pushes a sequence flag into every peer's flag slot.
//   k2: every rank waits for all flags, then each workgroup owns one
//       token row: read all N partials (7 remote xGMI reads), sum, add
//       residual (in-place update), RMSNorm with the layer weight, and
//       write the normed bf16 activation — the entire
//       all_reduce + add + rmsnorm sequence in two tiny kernels with no
//       host round trips, capturable inside hipGraphs (all state,
//       including the sequence number, lives on-device).
//
// Synchronization protocol (graph-replay safe — nothing is reset from
// the host): monotonically increasing device seq; double-buffered data
// slots by seq parity. Stream-serialized k1/k2 per rank make the
// slot-reuse distance 2 calls, and a rank can only start call s+2 after
// every peer finished call s (flag dependency), so slots never overlap.
// Spin-waits are bounded; on timeout the kernel records an error the
// host can query (xgmi_error_count) instead of wedging the GPU.
#include <torch/extension.h>

#include "common.h"

namespace {

constexpr int kMaxWorld = 8;
constexpr int64_t kMaxBytes = 8ll << 20;  // per slot; > 8 MiB falls back
constexpr int kFlagsOff = 2 * (kMaxBytes / 8);  // in uint64 units

struct CommState {
  void* bufs[kMaxWorld] = {nullptr};  // [rank] -> base of rank's buffer
  unsigned long long* seq = nullptr;  // device counter (local)
  unsigned int* err = nullptr;        // device error counter (local)
  int rank = -1;
  int world = 0;
  bool ready = false;
};
CommState g_comm;

// buffer layout (uint64 units): [slot0 | slot1 | flags[kMaxWorld] | k1_ctr]
DEV_INLINE unsigned long long* flags_of(void* base) {
  return reinterpret_cast<unsigned long long*>(base) + kFlagsOff;
}
DEV_INLINE unsigned int* k1_ctr_of(void* base) {
  return reinterpret_cast<unsigned int*>(flags_of(base) + kMaxWorld);
}
DEV_INLINE ushort* slot_of(void* base, unsigned long long s) {
  return reinterpret_cast<ushort*>(base) + (s & 1) * (kMaxBytes / 2);
}

struct Ptrs {
  void* p[kMaxWorld];
};

// ---- k1: publish my partial + push flags ------------------------------
__global__ void xgmi_publish_kernel(
    Ptrs bufs, const ushort* __restrict__ x, const int64_t n,
    const int rank, const int world, unsigned long long* seq) {
  const unsigned long long s = *seq;  // stable: bumped only at k2 end
  ushort* dst = slot_of(bufs.p[rank], s);
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  // 8-byte vector copy main body + scalar tail
  const int64_t n4 = n / 4;
  const uint64_t* src4 = reinterpret_cast<const uint64_t*>(x);
  uint64_t* dst4 = reinterpret_cast<uint64_t*>(dst);
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n4; i += stride)
    dst4[i] = src4[i];
  for (int64_t i = n4 * 4 + blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    dst[i] = x[i];
  // last workgroup (after ALL copies landed) pushes the flag to peers.
  // system-scope fence in EVERY workgroup: peer GPUs must see this wg's
  // slot writes before any flag does (per-XCD L2s on MI355X)
  __threadfence_system();
  __syncthreads();
  if (threadIdx.x == 0) {
    unsigned int* ctr = k1_ctr_of(bufs.p[rank]);
    if (atomicAdd(ctr, 1u) == gridDim.x - 1) {
      *ctr = 0u;
      __threadfence_system();  // data visible before any peer sees the flag
#pragma unroll
      for (int r = 0; r < kMaxWorld; ++r) {
        if (r < world) {
          volatile unsigned long long* f = flags_of(bufs.p[r]) + rank;
          *f = s + 1;
        }
      }
    }
  }
}

// ---- k2: wait, reduce, +residual, rmsnorm -----------------------------
// one workgroup per token row; H <= 16384 (32 f32 per thread at 512 thr)
template <int VPT>  // values per thread
__global__ void xgmi_reduce_norm_kernel(
    Ptrs bufs, ushort* __restrict__ out,        // [T, H] normed bf16
    ushort* __restrict__ residual,              // [T, H] in-place update
    const ushort* __restrict__ weight,          // [H]
    const float eps, const int H, const int rank, const int world,
    unsigned long long* seq, unsigned int* err) {
  const unsigned long long s = *seq;
  // every workgroup waits until all ranks published (bounded spin)
  {
    volatile unsigned long long* f = flags_of(bufs.p[rank]);
    if (threadIdx.x < kMaxWorld) {
      long long spins = 0;
      if ((int)threadIdx.x < world) {
        while (f[threadIdx.x] < s + 1) {
          __builtin_amdgcn_s_sleep(8);
          if (++spins > (1ll << 28)) {  // ~seconds: give up, flag error
            atomicAdd(err, 1u);
            break;
          }
        }
      }
    }
    __syncthreads();
    __threadfence_system();  // acquire: peer data after their flags
  }
  const int t = blockIdx.x;
  const int64_t row = (int64_t)t * H;
  float v[VPT];
  float sq = 0.f;
  const int nthr = blockDim.x;
#pragma unroll
  for (int j = 0; j < VPT; ++j) {
    const int c = threadIdx.x + j * nthr;
    if (c < H) {
      float acc = 0.f;
      for (int r = 0; r < world; ++r) {
        const ushort* src = slot_of(bufs.p[r], s);
        acc += bf16_to_f32(src[row + c]);
      }
      acc += bf16_to_f32(residual[row + c]);
      residual[row + c] = f32_to_bf16(acc);  // residual <- sum + residual
      v[j] = acc;
      sq += acc * acc;
    }
  }
  // block reduction of sum of squares
  __shared__ float red[16];
  sq = wave_reduce_sum(sq);
  const int wave = threadIdx.x / WAVE_SIZE;
  const int lane = threadIdx.x % WAVE_SIZE;
  if (lane == 0) red[wave] = sq;
  __syncthreads();
  const int nwaves = nthr / WAVE_SIZE;
  if (wave == 0) {
    float tot = (lane < nwaves) ? red[lane] : 0.f;
    tot = wave_reduce_sum(tot);
    if (lane == 0) red[0] = tot;
  }
  __syncthreads();
  const float rrms = rsqrtf(red[0] / H + eps);
#pragma unroll
  for (int j = 0; j < VPT; ++j) {
    const int c = threadIdx.x + j * nthr;
    if (c < H)
      out[row + c] =
          f32_to_bf16(v[j] * rrms * bf16_to_f32(weight[c]));
  }
  // last workgroup advances the device sequence number
  __syncthreads();
  if (threadIdx.x == 0) {
    unsigned int* ctr = k1_ctr_of(bufs.p[rank]);  // reused: k2 counter
    if (atomicAdd(ctr, 1u) == gridDim.x - 1) {
      *ctr = 0u;
      __threadfence();
      *seq = s + 1;
    }
  }
}

}  // namespace

// ---------------------------------------------------------------------
// host API

int64_t xgmi_buffer_bytes() {
  return 2 * kMaxBytes + (kMaxWorld + 2) * 8;
}

int64_t xgmi_max_elems() { return kMaxBytes / 2; }

// allocate my buffer, return the 64-byte hipIpcMemHandle_t as bytes
pybind11::bytes xgmi_alloc() {
  TORCH_CHECK(g_comm.bufs[0] == nullptr || !g_comm.ready,
              "xgmi comm already initialized");
  void* p = nullptr;
  TORCH_CHECK(hipMalloc(&p, xgmi_buffer_bytes()) == hipSuccess,
              "xgmi_alloc: hipMalloc failed");
  TORCH_CHECK(hipMemset(p, 0, xgmi_buffer_bytes()) == hipSuccess);
  g_comm.bufs[0] = p;  // temporarily park mine at [0]; connect() reorders
  hipIpcMemHandle_t h;
  TORCH_CHECK(hipIpcGetMemHandle(&h, p) == hipSuccess,
              "hipIpcGetMemHandle failed (needs dmabuf IPC; "
              "HSA_ENABLE_IPC_MODE_LEGACY=0)");
  return pybind11::bytes(reinterpret_cast<const char*>(&h), sizeof(h));
}

void xgmi_connect(int64_t rank, int64_t world,
                  const std::vector<pybind11::bytes>& handles) {
  TORCH_CHECK(world >= 2 && world <= kMaxWorld, "world out of range");
  TORCH_CHECK((int64_t)handles.size() == world, "need one handle per rank");
  void* mine = g_comm.bufs[0];
  TORCH_CHECK(mine != nullptr, "call xgmi_alloc first");
  for (int r = 0; r < world; ++r) {
    if (r == (int)rank) {
      g_comm.bufs[r] = mine;
      continue;
    }
    std::string hb = handles[r];
    TORCH_CHECK(hb.size() == sizeof(hipIpcMemHandle_t), "bad handle size");
    hipIpcMemHandle_t h;
    memcpy(&h, hb.data(), sizeof(h));
    void* p = nullptr;
    TORCH_CHECK(hipIpcOpenMemHandle(&p, h, hipIpcMemLazyEnablePeerAccess) ==
                    hipSuccess,
                "hipIpcOpenMemHandle failed for rank ", r);
    g_comm.bufs[r] = p;
  }
  TORCH_CHECK(hipMalloc(&g_comm.seq, 8) == hipSuccess);
  TORCH_CHECK(hipMemset(g_comm.seq, 0, 8) == hipSuccess);
  TORCH_CHECK(hipMalloc(&g_comm.err, 4) == hipSuccess);
  TORCH_CHECK(hipMemset(g_comm.err, 0, 4) == hipSuccess);
  g_comm.rank = (int)rank;
  g_comm.world = (int)world;
  g_comm.ready = true;
}

void xgmi_fused_allreduce_add_rmsnorm(torch::Tensor x, torch::Tensor residual,
                                      torch::Tensor weight, double eps) {
  TORCH_CHECK(g_comm.ready, "xgmi comm not connected");
  TORCH_CHECK(x.is_contiguous() && residual.is_contiguous());
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16);
  const int64_t n = x.numel();
  TORCH_CHECK(n * 2 <= kMaxBytes, "tensor exceeds one-shot buffer");
  const int H = x.size(-1);
  const int64_t T = n / H;
  TORCH_CHECK(H % 2 == 0 && H <= 16384);
  Ptrs ptrs;
  for (int r = 0; r < kMaxWorld; ++r) ptrs.p[r] = g_comm.bufs[r];
  auto stream = c10::hip::getCurrentHIPStream().stream();
  const int copy_blocks = (int)std::min<int64_t>(512, (n / 4 + 255) / 256 + 1);
  hipLaunchKernelGGL(xgmi_publish_kernel, dim3(copy_blocks), dim3(256), 0,
                     stream, ptrs, (const ushort*)x.data_ptr(), n,
                     g_comm.rank, g_comm.world, g_comm.seq);
  HIP_CHECK_KERNEL();
  const int nthr = 512;
#define LAUNCH_K2(VPT)                                                      \
  hipLaunchKernelGGL((xgmi_reduce_norm_kernel<VPT>), dim3((uint32_t)T),     \
                     dim3(nthr), 0, stream, ptrs, (ushort*)x.data_ptr(),    \
                     (ushort*)residual.data_ptr(),                          \
                     (const ushort*)weight.data_ptr(), (float)eps, H,       \
                     g_comm.rank, g_comm.world, g_comm.seq, g_comm.err)
  if (H <= nthr * 4) {
    LAUNCH_K2(4);
  } else if (H <= nthr * 8) {
    LAUNCH_K2(8);
  } else if (H <= nthr * 16) {
    LAUNCH_K2(16);
  } else {
    LAUNCH_K2(32);
  }
#undef LAUNCH_K2
  HIP_CHECK_KERNEL();
}

int64_t xgmi_error_count() {
  if (!g_comm.ready) return 0;
  unsigned int v = 0;
  hipMemcpy(&v, g_comm.err, 4, hipMemcpyDeviceToHost);
  return (int64_t)v;
}

void xgmi_shutdown() {
  if (!g_comm.ready && g_comm.bufs[0] == nullptr) return;
  for (int r = 0; r < kMaxWorld; ++r) {
    if (g_comm.bufs[r] == nullptr) continue;
    if (r == g_comm.rank || (!g_comm.ready && r == 0)) {
      hipFree(g_comm.bufs[r]);
    } else {
      hipIpcCloseMemHandle(g_comm.bufs[r]);
    }
    g_comm.bufs[r] = nullptr;
  }
  if (g_comm.seq) hipFree(g_comm.seq);
  if (g_comm.err) hipFree(g_comm.err);
  g_comm = CommState{};
}
