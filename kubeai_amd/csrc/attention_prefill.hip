// Paged flash-attention prefill (causal, GQA, prefix-cache aware) for gfx950.
//
// Semantics: kubeai_amd/ops/ref.py::paged_attention_prefill — new (chunk)
// query tokens attend to the full paged KV cache (which already contains
// this chunk's K/V via reshape_and_cache).
//
// MI355X-first structure (guide §3/§5/§6):
//  - workgroup = (seq, kv_head, 32-row q-tile); one wave per q-head of the
//    GQA group (G waves), so the staged KV tile in LDS is consumed by all G
//    heads — each KV byte crosses HBM once per 32*G query rows.
//  - each wave computes TWO 16-row MFMA sub-tiles per staged KV tile
//    (QBLK=32): doubles arithmetic intensity over the v1 16-row tile.
//  - QK^T and PV on v_mfma_f32_16x16x32_bf16; fp32 accumulation in AGPRs.
//  - K tile [32][128] and transposed V tile [128][32] staged in LDS with
//    the XOR swizzle (byte ^= (row&7)<<4) -> conflict-free ds_read_b128
//    B-fragment reads (guide §6 Guideline 4).
//  - online softmax entirely in C-fragment registers; the row (=q) direction
//    lives in the low 4 lane bits, so row max/sum are 4-step shfl_xor
//    butterflies; P goes through a swizzled LDS round-trip to reach
//    A-fragment layout for PV.
//
// Fragment layouts (verified by tests/test_kernels_gpu.py numerics):
//   A[M=16][K=32]: lane l holds A[l&15][(l>>4)*8 + j], j=0..7 (bf16)
//   B[K=32][N=16]: lane l holds B[(l>>4)*8 + j][l&15]
//   C[M=16][N=16]: lane l holds C[(l>>4)*4 + i][l&15], i=0..3 (fp32)
#include <torch/extension.h>

#include "common.h"

namespace {

constexpr int kBS = 16;    // cache block size
constexpr int kQB = 16;    // query rows per tile (32 costs occupancy: 201 VGPR -> 1 wave/SIMD, net loss)
constexpr int kMT = kQB / 16;
constexpr int kKVB = 32;   // kv tokens per tile (2 cache blocks)

typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

DEV_INLINE int swz(int row, int byte_off) {
  return byte_off ^ ((row & 7) << 4);
}

// CT = cache element type: ushort (bf16) or unsigned char (fp8 e5m2,
// converted to bf16 while staging — LDS layout and MFMA path unchanged)
// HD = head dim: 128 (llama family) or 256 (gemma2; K-chunk and
// PV-column loops scale with HD/32 and HD/16).
// SOFTCAP: gemma2 attention-logit capping s = cap*tanh(s/cap), applied
// after scale, before masking (0 disables; passed at runtime).
template <int G, typename CT = ushort, int HD = 128, bool CAP = false>
__launch_bounds__(G * WAVE_SIZE) __global__ void paged_prefill_kernel(
    ushort* __restrict__ out,            // [Tq, n_q, hd]
    const ushort* __restrict__ q,        // [Tq, n_q, hd]
    const CT* __restrict__ k_cache,      // [nb, n_kv, bs, hd]
    const CT* __restrict__ v_cache,
    const int32_t* __restrict__ block_tables,     // [B, max_blocks]
    const int32_t* __restrict__ query_start_loc,  // [B+1]
    const int32_t* __restrict__ seq_lens,         // [B]
    const float scale, const float softcap, const int window,
    const int n_kv, const int max_blocks,
    const int64_t q_stride) {
  const int b = blockIdx.x;
  const int kh = blockIdx.y;
  const int qtile = blockIdx.z;
  const int n_q = n_kv * G;
  const int wave = threadIdx.x / WAVE_SIZE;  // = q head within group
  const int lane = threadIdx.x % WAVE_SIZE;
  const int head = kh * G + wave;

  const int s0 = query_start_loc[b];
  const int q_len = query_start_loc[b + 1] - s0;
  if (qtile * kQB >= q_len) return;  // whole workgroup exits together
  const int L = seq_lens[b];
  const int ctx = L - q_len;  // absolute position of query row 0

  // rows this tile covers (clamped; invalid rows masked, never written)
  const int row_lo = qtile * kQB;
  const int n_rows = min(kQB, q_len - row_lo);
  // causal kv limit for this tile: last row's abs position
  const int kv_limit = ctx + row_lo + n_rows;  // exclusive
  const int n_kv_tiles = (kv_limit + kKVB - 1) / kKVB;

  __shared__ ushort k_lds[kKVB * HD];          // [tok][hd], swizzled
  __shared__ ushort v_lds[HD * kKVB];          // [hd][tok], swizzled
  __shared__ ushort p_lds[G][16 * kKVB];        // [row][tok], swizzled

  // ---- load Q fragments (2 sub-tiles x 4 K-chunks) from global ----
  bf16x8 q_frag[kMT][HD / 32];
#pragma unroll
  for (int mt = 0; mt < kMT; ++mt) {
    const int qrow = min(row_lo + mt * 16 + (lane & 15), q_len - 1);
    const ushort* qp =
        q + (int64_t)(s0 + qrow) * q_stride + (int64_t)head * HD;
#pragma unroll
    for (int kc = 0; kc < HD / 32; ++kc) {
      q_frag[mt][kc] =
          *reinterpret_cast<const bf16x8*>(qp + kc * 32 + (lane >> 4) * 8);
    }
  }

  f32x4 o_acc[kMT][HD / 16];
#pragma unroll
  for (int mt = 0; mt < kMT; ++mt)
#pragma unroll
    for (int c = 0; c < HD / 16; ++c) o_acc[mt][c] = {0.f, 0.f, 0.f, 0.f};
  float m_run[kMT][4], l_run[kMT][4];
#pragma unroll
  for (int mt = 0; mt < kMT; ++mt)
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      m_run[mt][i] = -INFINITY;
      l_run[mt][i] = 0.f;
    }

  const int32_t* bt = block_tables + (int64_t)b * max_blocks;

  // sliding window: the smallest qpos in this tile is ctx+row_lo, so KV
  // tiles ending at or before (ctx+row_lo - window) are masked for every
  // row and can be skipped outright
  const int kt0 =
      (window > 0) ? max(0, ctx + row_lo - window + 1) / kKVB : 0;

  for (int kt = kt0; kt < n_kv_tiles; ++kt) {
    const int kv_start = kt * kKVB;
    const int kv_valid = min(kKVB, kv_limit - kv_start);  // tokens staged

    // ---- cooperative staging: K -> k_lds, V -> v_lds transposed ----
    __syncthreads();  // previous tile fully consumed
    {
      const int nvec = kKVB * HD / 8;
      for (int i = threadIdx.x; i < nvec; i += G * WAVE_SIZE) {
        const int tok = i / (HD / 8);
        const int col8 = i % (HD / 8);
        if (tok >= kv_valid) {
          // zero unstaged V: P rows are 0 there, but 0 * stale-NaN would
          // poison the PV MFMA accumulator (K can stay stale: scores for
          // invalid tokens are forced to -inf before softmax)
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            const int hdrow = col8 * 8 + j;
            *reinterpret_cast<ushort*>(
                reinterpret_cast<char*>(v_lds) +
                swz(hdrow, (hdrow * kKVB + tok) * 2)) = 0;
          }
        }
        if (tok < kv_valid) {
          const int abs_tok = kv_start + tok;
          const int64_t blk = bt[abs_tok / kBS];
          const CT* src = k_cache +
                          (((blk * n_kv + kh) * kBS) + abs_tok % kBS) * HD;
          const CT* vsrc = v_cache +
                           (((blk * n_kv + kh) * kBS) + abs_tok % kBS) * HD;
          bf16x8 kk, vv;
          if constexpr (sizeof(CT) == 2) {
            kk = *reinterpret_cast<const bf16x8*>(src + col8 * 8);
            vv = *reinterpret_cast<const bf16x8*>(vsrc + col8 * 8);
          } else {
            // fp8: 8-byte loads, packed-convert e5m2 -> bf16 in-register
            uint64_t kraw = *reinterpret_cast<const uint64_t*>(src + col8 * 8);
            uint64_t vraw = *reinterpret_cast<const uint64_t*>(vsrc + col8 * 8);
            const ushort8 kc = e5m2x8_to_bf16x8(kraw);
            const ushort8 vc = e5m2x8_to_bf16x8(vraw);
#pragma unroll
            for (int j = 0; j < 8; ++j) {
              kk[j] = (short)kc[j];
              vv[j] = (short)vc[j];
            }
          }
          // K: 16B vector write, swizzled row=tok
          *reinterpret_cast<bf16x8*>(
              reinterpret_cast<char*>(k_lds) +
              swz(tok, tok * HD * 2 + col8 * 16)) = kk;
          // V: scatter-transpose 8 elems (row=hd, col=tok)
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            const int hdrow = col8 * 8 + j;
            *reinterpret_cast<ushort*>(
                reinterpret_cast<char*>(v_lds) +
                swz(hdrow, (hdrow * kKVB + tok) * 2)) = (ushort)vv[j];
          }
        }
      }
    }
    __syncthreads();

#pragma unroll
    for (int mt = 0; mt < kMT; ++mt) {
      const int mrow_lo = row_lo + mt * 16;
      if (mt * 16 >= n_rows) break;               // sub-tile fully past q_len
      if (kv_start > ctx + mrow_lo + 15) continue; // fully causal-masked

      // ---- S = Q.K^T over two 16-col subtiles ----
      f32x4 s_frag[2];
#pragma unroll
      for (int nsub = 0; nsub < 2; ++nsub) {
        s_frag[nsub] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int kc = 0; kc < HD / 32; ++kc) {
          const int tok = nsub * 16 + (lane & 15);
          const bf16x8 k_frag = *reinterpret_cast<const bf16x8*>(
              reinterpret_cast<const char*>(k_lds) +
              swz(tok, tok * HD * 2 + kc * 64 + (lane >> 4) * 16));
          s_frag[nsub] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              q_frag[mt][kc], k_frag, s_frag[nsub], 0, 0, 0);
        }
      }

      // ---- mask + online softmax in C-fragment layout ----
      float p[2][4];
      float corr[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int r = mt * 16 + (lane >> 4) * 4 + i;
        const int qpos = ctx + row_lo + r;
        float s0v = s_frag[0][i] * scale;
        float s1v = s_frag[1][i] * scale;
        if constexpr (CAP) {
          s0v = softcap * tanhf(s0v / softcap);
          s1v = softcap * tanhf(s1v / softcap);
        }
        const int kv0 = kv_start + (lane & 15);
        const int kv1 = kv0 + 16;
        if (kv0 > qpos || r >= n_rows) s0v = -INFINITY;
        if (kv1 > qpos || r >= n_rows) s1v = -INFINITY;
        if (window > 0) {  // sliding window: keys in (qpos-window, qpos]
          if (kv0 <= qpos - window) s0v = -INFINITY;
          if (kv1 <= qpos - window) s1v = -INFINITY;
        }
        float rmax = fmaxf(s0v, s1v);
#pragma unroll
        for (int off = 8; off > 0; off >>= 1)
          rmax = fmaxf(rmax, __shfl_xor(rmax, off, 64));
        const float m_new = fmaxf(m_run[mt][i], rmax);
        corr[i] =
            (m_run[mt][i] == -INFINITY) ? 0.f : __expf(m_run[mt][i] - m_new);
        p[0][i] = (s0v == -INFINITY) ? 0.f : __expf(s0v - m_new);
        p[1][i] = (s1v == -INFINITY) ? 0.f : __expf(s1v - m_new);
        float rsum = p[0][i] + p[1][i];
#pragma unroll
        for (int off = 8; off > 0; off >>= 1) rsum += __shfl_xor(rsum, off, 64);
        l_run[mt][i] = l_run[mt][i] * corr[i] + rsum;
        m_run[mt][i] = m_new;
      }
#pragma unroll
      for (int c = 0; c < HD / 16; ++c)
#pragma unroll
        for (int i = 0; i < 4; ++i) o_acc[mt][c][i] *= corr[i];

      // ---- P -> bf16 A-fragment via swizzled LDS round trip ----
#pragma unroll
      for (int nsub = 0; nsub < 2; ++nsub)
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          const int r16 = (lane >> 4) * 4 + i;  // row within the sub-tile
          const int tok = nsub * 16 + (lane & 15);
          *reinterpret_cast<ushort*>(
              reinterpret_cast<char*>(p_lds[wave]) +
              swz(r16, (r16 * kKVB + tok) * 2)) = f32_to_bf16(p[nsub][i]);
        }
      // wave-local LDS dependency; compiler inserts the lgkmcnt wait
      bf16x8 p_frag = *reinterpret_cast<const bf16x8*>(
          reinterpret_cast<const char*>(p_lds[wave]) +
          swz(lane & 15, ((lane & 15) * kKVB + (lane >> 4) * 8) * 2));

      // ---- O += P.V ----
#pragma unroll
      for (int c = 0; c < HD / 16; ++c) {
        const int hdcol = c * 16 + (lane & 15);
        const bf16x8 v_frag = *reinterpret_cast<const bf16x8*>(
            reinterpret_cast<const char*>(v_lds) +
            swz(hdcol, (hdcol * kKVB + (lane >> 4) * 8) * 2));
        o_acc[mt][c] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            p_frag, v_frag, o_acc[mt][c], 0, 0, 0);
      }
    }
  }

  // ---- write O / l ----
#pragma unroll
  for (int mt = 0; mt < kMT; ++mt)
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int r = mt * 16 + (lane >> 4) * 4 + i;
      if (r >= n_rows) continue;
      const float inv_l = 1.0f / l_run[mt][i];
      ushort* op = out + ((int64_t)(s0 + row_lo + r) * n_q + head) * HD;
#pragma unroll
      for (int c = 0; c < HD / 16; ++c)
        op[c * 16 + (lane & 15)] = f32_to_bf16(o_acc[mt][c][i] * inv_l);
    }
}

}  // namespace

// attention_prefill_v2.hip; returns false when the shape is unsupported
bool paged_attention_prefill_v2(torch::Tensor out, torch::Tensor q,
                                torch::Tensor k_cache, torch::Tensor v_cache,
                                torch::Tensor block_tables,
                                torch::Tensor query_start_loc,
                                torch::Tensor seq_lens, double scale);

void paged_attention_prefill(torch::Tensor out, torch::Tensor q,
                             torch::Tensor k_cache, torch::Tensor v_cache,
                             torch::Tensor block_tables,
                             torch::Tensor query_start_loc,
                             torch::Tensor seq_lens, double scale,
                             int64_t window, double softcap) {
  TORCH_CHECK(q.stride(2) == 1 && q.stride(1) == q.size(2));
  TORCH_CHECK(out.is_contiguous());
  TORCH_CHECK(k_cache.is_contiguous() && v_cache.is_contiguous());
  TORCH_CHECK(q.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(k_cache.scalar_type() == torch::kBFloat16 ||
              k_cache.scalar_type() == torch::kFloat8_e5m2);
  TORCH_CHECK(block_tables.scalar_type() == torch::kInt32);
  TORCH_CHECK(query_start_loc.scalar_type() == torch::kInt32);
  TORCH_CHECK(seq_lens.scalar_type() == torch::kInt32);
  const int n_q = q.size(1), hd = q.size(2);
  const int n_kv = k_cache.size(1);
  const int B = seq_lens.size(0);
  const int max_blocks = block_tables.size(1);
  TORCH_CHECK(hd == 128 || hd == 256,
              "prefill kernel supports head_dim 128 or 256");
  TORCH_CHECK(k_cache.size(2) == kBS);
  const int G = n_q / n_kv;
  TORCH_CHECK(n_q % n_kv == 0);
  if (q.size(0) == 0) return;
  // v2: 8-wave 32x32-MFMA ladder (attention_prefill_v2.hip), G in
  // {1,2,4,8} — DEFAULT (1.5-1.8x v1 after the defer-max/VALU pass:
  // 268 TF vs 150 @Tq=8192, numerics cos=1.0 across the full isolation
  // matrix, profiles/r02_results.md). KUBEAI_PREFILL_V2=0 falls back.
  static const bool use_v2 = []() {
    const char* e = getenv("KUBEAI_PREFILL_V2");
    return e == nullptr || e[0] != '0';
  }();
  // the v2 ladder has no sliding-window masking (interior-tile fast
  // path assumes plain causal) — windowed models run the v1 kernel
  // the v2 ladder covers the plain-causal hd=128 fast path only
  if (window == 0 && softcap == 0.0 && hd == 128 && use_v2 &&
      paged_attention_prefill_v2(out, q, k_cache, v_cache, block_tables,
                                 query_start_loc, seq_lens, scale))
    return;
  const int Tq = q.size(0);
  const int n_qtiles_max = (Tq + kQB - 1) / kQB;  // per-seq early exit
  dim3 grid(B, n_kv, n_qtiles_max);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  const bool fp8_cache = k_cache.scalar_type() == torch::kFloat8_e5m2;
#define LAUNCH_CT_HD(GG, CT, HDV)                                          \
  do {                                                                      \
    if (softcap > 0.0)                                                      \
      LAUNCH_IMPL(GG, CT, HDV, true);                                       \
    else                                                                    \
      LAUNCH_IMPL(GG, CT, HDV, false);                                      \
  } while (0)
#define LAUNCH_IMPL(GG, CT, HDV, CAPV)                                      \
  hipLaunchKernelGGL((paged_prefill_kernel<GG, CT, HDV, CAPV>), grid,       \
                     dim3(GG * WAVE_SIZE), 0, stream,                       \
                     (ushort*)out.data_ptr(), (const ushort*)q.data_ptr(),  \
                     (const CT*)k_cache.data_ptr(),                         \
                     (const CT*)v_cache.data_ptr(),                         \
                     block_tables.data_ptr<int32_t>(),                      \
                     query_start_loc.data_ptr<int32_t>(),                   \
                     seq_lens.data_ptr<int32_t>(), (float)scale,            \
                     (float)softcap, (int)window, n_kv,                     \
                     max_blocks, q.stride(0))
#define LAUNCH(GG)                                                         \
  do {                                                                     \
    if (hd == 256) {                                                       \
      if (fp8_cache) LAUNCH_CT_HD(GG, unsigned char, 256);                  \
      else LAUNCH_CT_HD(GG, ushort, 256);                                   \
    } else if (fp8_cache) {                                                 \
      LAUNCH_CT_HD(GG, unsigned char, 128);                                 \
    } else {                                                                \
      LAUNCH_CT_HD(GG, ushort, 128);                                        \
    }                                                                       \
  } while (0)
  switch (G) {
    case 1: LAUNCH(1); break;
    case 2: LAUNCH(2); break;
    case 4: LAUNCH(4); break;
    case 8: LAUNCH(8); break;
    default: TORCH_CHECK(false, "unsupported GQA group size ", G);
  }
#undef LAUNCH
#undef LAUNCH_CT_HD
#undef LAUNCH_IMPL
  HIP_CHECK_KERNEL();
}
