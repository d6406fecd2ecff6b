// Paged flash-attention prefill v2 for gfx950 — the 8-wave 32x32 MFMA
// "ladder" structure (guide Appendix B, learn_hip m214: ~900 TF on this
// GQA D=128 shape in plain HIP), re-derived for the PAGED KV setting.
//
// Semantics: kubeai_amd/ops/ref.py::paged_attention_prefill (identical
// to attention_prefill.hip v1; v1 remains the fallback for odd GQA
// group sizes).
//
// Structure (per workgroup = (seq, kv_head, q-chunk), 8 waves):
//  - waves = G heads x (8/G) 32-row q subtiles; all 8 share each staged
//    64-token KV tile, so every KV byte crossing HBM feeds 8 * 32 q rows.
//  - swapped QK^T on v_mfma_f32_32x32x16_bf16: S^T = mfma(A=K, B=Q^T)
//    puts query in the C-fragment COLUMN (lane&31) — the entire online
//    softmax (row max, exp, row sum, m/l state) is per-lane scalar work
//    plus ONE cross-half shuffle; no LDS round trip for P.
//  - P -> PV A-fragments in-register: v_cvt_pk_bf16_f32 packs pairs,
//    one __shfl_xor(32) per quad exchanges the half-rows (T12 idiom).
//  - K tile [64][128] XOR-swizzled in LDS (conflict-free ds_read_b128);
//    V tile stored 4x16-SUBTILED and consumed with the hardware
//    transpose read ds_read_b64_tr_b16 (T10) — V stages with plain 16 B
//    vector writes (v1's scatter-transpose bank storm is gone).
//  - double-buffered LDS (2 x 32 KB); next tile's global loads issue
//    before this tile's compute (async-STAGE), land after the barrier.
//
// Fragment layouts (guide §3, device-verified there):
//   A[M=32][K=16]: lane l holds A[l&31][(l>>5)*8 + j], j=0..7
//   B[K=16][N=32]: lane l holds B[(l>>5)*8 + j][l&31]
//   C[M=32][N=32]: lane l holds C[(reg&3) + 8*(reg>>2) + 4*(l>>5)][l&31]
//   ds_read_b64_tr_b16: lane l elem j reads lds[(l&15) + j*16 + (l>>4)*64]
#include <torch/extension.h>

#include "common.h"

namespace {

constexpr int kHD = 128;
constexpr int kBS = 16;   // cache block size (tokens)
constexpr int kKVB = 64;  // kv tokens per staged tile (4 cache blocks)
constexpr int kQB = 32;   // q rows per wave

typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(16))) float f32x16;

DEV_INLINE int swz(int row, int byte_off) {
  return byte_off ^ ((row & 7) << 4);
}

DEV_INLINE uint32_t cvt_pk_bf16(float a, float b) {
  uint32_t r;
  asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(a), "v"(b));
  return r;
}

// C-fragment row for register index `reg` in lane-half `hi`
DEV_INLINE int crow(int reg, int hi) {
  return (reg & 3) + 8 * (reg >> 2) + 4 * hi;
}

// MINW: min waves per SIMD fed to launch_bounds — the VGPR budget knob.
// Forcing 4 waves/SIMD caps arch VGPRs at 128 and spills ~300 VGPRs +
// ~430 SGPRs (~650 B/lane scratch): 90 TF @Tq=8192. MINW=2 gives 256
// VGPRs, near-zero spill (≤21 v, ~80 s), 283 TF — 1 workgroup/CU beats
// 2 spilly ones 3x over. Measured same-box; KUBEAI_V2_OCC=4 re-checks.
// NW: waves per workgroup. 8 (default) = 256 q rows/WG, 1 WG/CU at 256
// VGPRs. 4 = 32*NW/G q rows/WG with HALF the arithmetic intensity per
// KV byte but TWO co-resident WGs/CU (128 KB LDS) to overlap each
// other's latency stalls — opt-in experiment via KUBEAI_V2_WAVES=4.
template <int G, typename CT = ushort, int DBG = 0, int MINW = 2,
          int NW = 8>
__launch_bounds__(NW * WAVE_SIZE, MINW) __global__
void paged_prefill_v2_kernel(
    ushort* __restrict__ out,            // [Tq, n_q, hd]
    const ushort* __restrict__ q,        // [Tq, n_q, hd]
    const CT* __restrict__ k_cache,      // [nb, n_kv, bs, hd]
    const CT* __restrict__ v_cache,
    const int32_t* __restrict__ block_tables,     // [B, max_blocks]
    const int32_t* __restrict__ query_start_loc,  // [B+1]
    const int32_t* __restrict__ seq_lens,         // [B]
    const float scale, const int n_kv, const int max_blocks,
    const int64_t q_stride) {
  // DBG bisect switches (host env KUBEAI_V2_DBG, compile-time so the
  // fast path carries no runtime branches or extra SGPR pressure):
  //   bit 0: synchronous single-slot staging (no async double buffer)
  //   bit 1: plain scalar V reads instead of ds_read_b64_tr_b16
  constexpr bool kSyncStage = DBG & 1;
  constexpr bool kSimpleV = DBG & 2;
  // NW < G would leave heads without waves; the host never launches
  // that combination (falls back to NW=8), and the max() here keeps the
  // dead instantiation compilable without a hard error
  constexpr int kQSUB = (NW >= G) ? NW / G : 1;  // q subtiles per head
  constexpr int kQROWS = kQSUB * kQB;   // q rows per workgroup
  const int b = blockIdx.x;
  const int kh = blockIdx.y;
  const int qtile = blockIdx.z;
  const int n_q = n_kv * G;
  const int wave = threadIdx.x / WAVE_SIZE;
  const int lane = threadIdx.x % WAVE_SIZE;
  const int g = wave / kQSUB;           // head within GQA group
  const int qs = wave % kQSUB;          // q subtile of this wave
  const int head = kh * G + g;
  const int hi = lane >> 5;
  const int ln31 = lane & 31;

  const int s0 = query_start_loc[b];
  const int q_len = query_start_loc[b + 1] - s0;
  if (qtile * kQROWS >= q_len) return;  // uniform workgroup exit
  const int L = seq_lens[b];
  const int ctx = L - q_len;

  const int row_lo = qtile * kQROWS;            // workgroup's first row
  const int n_rows = min(kQROWS, q_len - row_lo);
  const int kv_limit = ctx + row_lo + n_rows;   // exclusive
  const int n_tiles = (kv_limit + kKVB - 1) / kKVB;

  // my wave's rows: [wrow0, wrow0+32) relative to row_lo
  const int wrow0 = qs * kQB;
  const int my_q = wrow0 + ln31;                // row within workgroup
  const int qpos = ctx + row_lo + my_q;         // absolute position
  const bool row_ok = my_q < n_rows;

  // LDS: K row-major swizzled; V 4x16-subtiled for tr_b16 reads
  __shared__ ushort k_lds[2][kKVB * kHD];
  __shared__ ushort v_lds[2][kKVB * kHD];

  // ---- Q in registers: B-fragment per 16-wide hd chunk --------------
  bf16x8 q_frag[8];
  {
    const int qrow = min(row_lo + my_q, q_len - 1);
    const ushort* qp =
        q + (int64_t)(s0 + qrow) * q_stride + (int64_t)head * kHD;
#pragma unroll
    for (int c = 0; c < 8; ++c)
      q_frag[c] =
          *reinterpret_cast<const bf16x8*>(qp + c * 16 + hi * 8);
  }

  f32x16 o_acc[4];
#pragma unroll
  for (int nb = 0; nb < 4; ++nb)
#pragma unroll
    for (int r = 0; r < 16; ++r) o_acc[nb][r] = 0.f;
  float m_run = -INFINITY, l_run = 0.f;

  const int32_t* bt = block_tables + (int64_t)b * max_blocks;

  // ---- cooperative staging: 2 K + 2 V 8-element units per thread ----
  // unit u of 1024: tok = u>>4, c8 = u&15 (16 bf16-8 units per 128-row).
  // All thread-invariant addressing is hoisted; per tile only the block
  // lookup + validity remain.
  constexpr int NSTG = (kKVB * kHD / 8) / (NW * WAVE_SIZE);
  bf16x8 st_k[NSTG], st_v[NSTG];
  bool st_ok[NSTG];
  int h_tok[NSTG], h_coff[NSTG], h_kdst[NSTG], h_vdst[NSTG];
#pragma unroll
  for (int i = 0; i < NSTG; ++i) {
    const int u = threadIdx.x + i * NW * WAVE_SIZE;
    const int tok = u >> 4;
    const int c8 = u & 15;
    h_tok[i] = tok;
    h_coff[i] = c8 * 8;
    h_kdst[i] = swz(tok, tok * kHD * 2 + c8 * 16);
    h_vdst[i] = ((tok >> 2) * 8 + (c8 >> 1)) * 64 + (tok & 3) * 16 +
                (c8 & 1) * 8;
  }
  auto issue_tile_loads = [&](int kt) {
    const int kv_start = kt * kKVB;
    const int kv_valid = min(kKVB, kv_limit - kv_start);
#pragma unroll
    for (int i = 0; i < NSTG; ++i) {
      const int tok = h_tok[i];
      st_ok[i] = tok < kv_valid;
      if (st_ok[i]) {
        const int abs_tok = kv_start + tok;
        const int64_t blk = bt[abs_tok / kBS];
        const int64_t base =
            (((blk * n_kv + kh) * kBS) + abs_tok % kBS) * kHD + h_coff[i];
        if constexpr (sizeof(CT) == 2) {
          st_k[i] = *reinterpret_cast<const bf16x8*>(k_cache + base);
          st_v[i] = *reinterpret_cast<const bf16x8*>(v_cache + base);
        } else {
          const uint64_t kraw =
              *reinterpret_cast<const uint64_t*>(k_cache + base);
          const uint64_t vraw =
              *reinterpret_cast<const uint64_t*>(v_cache + base);
          const ushort8 kc = e5m2x8_to_bf16x8(kraw);
          const ushort8 vc = e5m2x8_to_bf16x8(vraw);
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            st_k[i][j] = (short)kc[j];
            st_v[i][j] = (short)vc[j];
          }
        }
      } else {
        // V must be zero where P is masked (0 * stale-NaN poisons PV);
        // K can stay stale (scores forced to -inf)
#pragma unroll
        for (int j = 0; j < 8; ++j) st_v[i][j] = 0;
      }
    }
  };
  auto write_tile = [&](int buf) {
#pragma unroll
    for (int i = 0; i < NSTG; ++i) {
      // K: swizzled row-major (skip stale-write cost only when invalid)
      if (st_ok[i])
        *reinterpret_cast<bf16x8*>(
            reinterpret_cast<char*>(k_lds[buf]) + h_kdst[i]) = st_k[i];
      // V: subtiled; invalid units were zeroed at load
      *reinterpret_cast<bf16x8*>(&v_lds[buf][h_vdst[i]]) = st_v[i];
    }
  };

  if constexpr (!kSyncStage) {
    issue_tile_loads(0);
    write_tile(0);
    __syncthreads();
  }

  for (int kt = 0; kt < n_tiles; ++kt) {
    const int cur = kSyncStage ? 0 : (kt & 1);
    const int kv_start = kt * kKVB;
    const int kv_valid = min(kKVB, kv_limit - kv_start);
    const bool have_next = !kSyncStage && kt + 1 < n_tiles;
    if constexpr (kSyncStage) {
      __syncthreads();  // everyone done with the previous tile
      issue_tile_loads(kt);
      write_tile(0);
      __syncthreads();
    }
    if (have_next) issue_tile_loads(kt + 1);

    // wave-level skip: this wave's rows all causally precede the tile
    const int wave_kv_hi = ctx + row_lo + min(wrow0 + kQB, n_rows) - 1;
    if (kv_start <= wave_kv_hi) {
      // ---- S^T = K . Q^T over two 32-kv subtiles -------------------
      __builtin_amdgcn_s_setprio(1);  // T9: keep MFMA issue ahead
      f32x16 st2[2];
#pragma unroll
      for (int sub = 0; sub < 2; ++sub) {
#pragma unroll
        for (int r = 0; r < 16; ++r) st2[sub][r] = 0.f;
#pragma unroll
        for (int c = 0; c < 8; ++c) {
          const int tok = sub * 32 + ln31;
          const bf16x8 k_frag = *reinterpret_cast<const bf16x8*>(
              reinterpret_cast<const char*>(k_lds[cur]) +
              swz(tok, tok * kHD * 2 + c * 32 + hi * 16));
          st2[sub] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              k_frag, q_frag[c], st2[sub], 0, 0, 0);
        }
      }

      __builtin_amdgcn_s_setprio(0);
      // ---- mask + per-lane online softmax (q = column = ln31) ------
      // interior tiles (fully inside every row's causal window, full
      // kv_valid, all rows live) skip the per-element mask entirely —
      // wave-uniform, so it also drops the compare/select chains
      const bool tile_interior =
          kv_valid == kKVB && wrow0 + kQB <= n_rows &&
          kv_start + kKVB - 1 <= ctx + row_lo + wrow0;
      float pmax = -INFINITY;
      if (tile_interior) {
#pragma unroll
        for (int sub = 0; sub < 2; ++sub)
#pragma unroll
          for (int r = 0; r < 16; ++r) {
            const float s = st2[sub][r] * scale;
            st2[sub][r] = s;
            pmax = fmaxf(pmax, s);
          }
      } else {
        // single merged bound: kv is allowed iff kv <= kv_hi
        const int kv_hi =
            row_ok ? min(qpos, kv_start + kv_valid - 1) : -1;
#pragma unroll
        for (int sub = 0; sub < 2; ++sub)
#pragma unroll
          for (int r = 0; r < 16; ++r) {
            const int kv = kv_start + sub * 32 + crow(r, hi);
            float s = st2[sub][r] * scale;
            if (kv > kv_hi) s = -INFINITY;
            st2[sub][r] = s;
            pmax = fmaxf(pmax, s);
          }
      }
      pmax = fmaxf(pmax, __shfl_xor(pmax, 32, 64));  // merge lane halves
      // defer-max (T ladder): skip the whole rescale path while no
      // row's max grows by more than 8 (exp stays bounded by e^8) —
      // wave-uniform so the cross-lane corr gather can be skipped too
      const bool defer = __all(pmax <= m_run + 8.f);
      float m_new = m_run;
      if (!defer) {
        m_new = fmaxf(m_run, pmax);
        const float corr =
            (m_run == -INFINITY) ? 0.f : __expf(m_run - m_new);
        l_run *= corr;
        // o_acc rows are the C-fragment rows crow(r, hi) — NOT this
        // lane's softmax row (q = ln31). Rescale each accumulator row
        // by ITS OWN correction, gathered from its owning lane.
        float corr_row[16];
#pragma unroll
        for (int r = 0; r < 16; ++r)
          corr_row[r] = __shfl(corr, (lane & 32) + crow(r, hi), 64);
#pragma unroll
        for (int nb = 0; nb < 4; ++nb)
#pragma unroll
          for (int r = 0; r < 16; ++r) o_acc[nb][r] *= corr_row[r];
      }
      float rsum = 0.f;
      if (tile_interior) {
#pragma unroll
        for (int sub = 0; sub < 2; ++sub)
#pragma unroll
          for (int r = 0; r < 16; ++r) {
            const float p = __expf(st2[sub][r] - m_new);
            st2[sub][r] = p;
            rsum += p;
          }
      } else {
#pragma unroll
        for (int sub = 0; sub < 2; ++sub)
#pragma unroll
          for (int r = 0; r < 16; ++r) {
            const float p =
                (st2[sub][r] == -INFINITY) ? 0.f : __expf(st2[sub][r] - m_new);
            st2[sub][r] = p;
            rsum += p;
          }
      }
      rsum += __shfl_xor(rsum, 32, 64);
      l_run += rsum;
      m_run = m_new;

      // ---- P -> A fragments (in-register, T12) ---------------------
      // pa[ks]: lane holds P[q=ln31][ks*16 + hi*8 + jj], jj=0..7.
      // quad jj<4 comes from half-0's regs [4*q8..], jj>=4 from half-1's
      // (q8 = (ks&1)*2 + hi of the RECEIVER); each lane packs the quad
      // its partner needs and one shfl_xor(32) swaps them.
      bf16x8 pa[4];
#pragma unroll
      for (int ks = 0; ks < 4; ++ks) {
        const int sub = ks >> 1;
        // my own quad (source half == my half): regs 4*q8_own..+3
        const int q8_own = (ks & 1) * 2 + hi;
        const uint32_t own0 =
            cvt_pk_bf16(st2[sub][4 * q8_own], st2[sub][4 * q8_own + 1]);
        const uint32_t own1 =
            cvt_pk_bf16(st2[sub][4 * q8_own + 2], st2[sub][4 * q8_own + 3]);
        // quad my partner needs: its q8 = (ks&1)*2 + (1-hi)
        const int q8_par = (ks & 1) * 2 + 1 - hi;
        const uint32_t snd0 =
            cvt_pk_bf16(st2[sub][4 * q8_par], st2[sub][4 * q8_par + 1]);
        const uint32_t snd1 =
            cvt_pk_bf16(st2[sub][4 * q8_par + 2], st2[sub][4 * q8_par + 3]);
        const uint32_t got0 = __shfl_xor((int)snd0, 32, 64);
        const uint32_t got1 = __shfl_xor((int)snd1, 32, 64);
        // assemble: jj<4 = half-0 source, jj>=4 = half-1 source
        uint32_t w0 = hi ? got0 : own0;
        uint32_t w1 = hi ? got1 : own1;
        uint32_t w2 = hi ? own0 : got0;
        uint32_t w3 = hi ? own1 : got1;
        uint32_t* pw = reinterpret_cast<uint32_t*>(&pa[ks]);
        pw[0] = w0; pw[1] = w1; pw[2] = w2; pw[3] = w3;
      }

      // ---- O += P.V via hardware-transpose V reads -----------------
      // ds_read_b64_tr_b16 semantics (device-decoded, dbg_v2 b3): a
      // 4x4 transpose over each 16-lane group's lane grid (row=l>>2,
      // col=l&3): result[l][j] = lds[addr_of_lane((l&~15) + ((l>>2)&3)
      // + 4*j) + 2*(l&3)]. Solving for the address each lane must
      // supply so lane l's elem jj = V[ks*16+(l>>5)*8+jj][nb*32+(l&31)]
      // in the 4x16-subtiled layout:
      //   byte addr(m) = ks*4096 + tr*1024 + (m>>5)*2048 + nb*256
      //     + ((m>>4)&1)*128 + ((m>>2)&3)*32 + (m&3)*8
      __builtin_amdgcn_s_setprio(1);  // PV MFMA cluster
      typedef __attribute__((address_space(3))) const char as3_char;
      as3_char* vbase =
          (as3_char*)(v_lds[cur]) +
          (hi * 2048 + ((lane >> 4) & 1) * 128 + ((lane >> 2) & 3) * 32 +
           (lane & 3) * 8);
#pragma unroll
      for (int nb = 0; nb < 4; ++nb) {
#pragma unroll
        for (int ks = 0; ks < 4; ++ks) {
          bf16x8 v_frag;
          if constexpr (kSimpleV) {
            // bisect mode: direct scalar reads from the subtiled layout
#pragma unroll
            for (int jj = 0; jj < 8; ++jj) {
              const int row = ks * 16 + hi * 8 + jj;
              const int col = nb * 32 + ln31;
              const int off = ((row >> 2) * 8 + (col >> 4)) * 64 +
                              (row & 3) * 16 + (col & 15);
              v_frag[jj] = (short)v_lds[cur][off];
            }
          } else {
            typedef __attribute__((ext_vector_type(4))) short bf16x4;
            bf16x4 lo4, hi4;
            as3_char* a0 = vbase + ks * 4096 + nb * 256;
            // "=&v" early-clobber: without it the allocator may alias an
            // output with the address VGPR — the first read's writeback
            // then races the second read's address consumption
            // (sporadic corruption, shape/schedule dependent)
            asm volatile(
                "ds_read_b64_tr_b16 %0, %2\n\t"
                "ds_read_b64_tr_b16 %1, %2 offset:1024\n\t"
                "s_waitcnt lgkmcnt(0)"
                : "=&v"(lo4), "=&v"(hi4)
                : "v"(a0));
#pragma unroll
            for (int j = 0; j < 4; ++j) {
              v_frag[j] = lo4[j];
              v_frag[4 + j] = hi4[j];
            }
          }
          o_acc[nb] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              pa[ks], v_frag, o_acc[nb], 0, 0, 0);
        }
      }
    }

    __builtin_amdgcn_s_setprio(0);
    // single barrier per tile: writing buf cur^1 cannot conflict with
    // concurrent reads of buf cur, and the previous iteration's barrier
    // already separated it from the last reads of cur^1
    if (have_next) write_tile(cur ^ 1);
    __syncthreads();  // staged tile visible before next iteration
  }

  // ---- normalize + write O ------------------------------------------
  // o_acc rows are crow(r, hi); l_run is indexed by q=ln31 — gather the
  // needed 1/l via 16 lane-indexed shuffles (once, not per tile)
  const float inv_l = (l_run > 0.f) ? 1.0f / l_run : 0.f;
  float inv_row[16];
#pragma unroll
  for (int r = 0; r < 16; ++r)
    inv_row[r] = __shfl(inv_l, (lane & 32) + crow(r, hi), 64);
  // row validity per reg: row = crow(r, hi) (+ wave base)
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int row = wrow0 + crow(r, hi);
    if (row >= n_rows) continue;
    ushort* op = out + ((int64_t)(s0 + row_lo + row) * n_q + head) * kHD;
#pragma unroll
    for (int nb = 0; nb < 4; ++nb)
      op[nb * 32 + ln31] = f32_to_bf16(o_acc[nb][r] * inv_row[r]);
  }
}

}  // namespace

bool paged_attention_prefill_v2(torch::Tensor out, torch::Tensor q,
                                torch::Tensor k_cache, torch::Tensor v_cache,
                                torch::Tensor block_tables,
                                torch::Tensor query_start_loc,
                                torch::Tensor seq_lens, double scale) {
  const int n_q = q.size(1), hd = q.size(2);
  const int n_kv = k_cache.size(1);
  const int G = n_q / n_kv;
  if (hd != kHD || k_cache.size(2) != kBS ||
      (G != 1 && G != 2 && G != 4 && G != 8))
    return false;  // caller falls back to v1
  const int B = query_start_loc.size(0) - 1;
  const int max_blocks = block_tables.size(1);
  // small-q chunked continuations underfill the 8-wave workgroups
  // (few z-tiles); v1's 16-row tiles win there (profiles/r02)
  if (q.size(0) <= 1024 && B == 1) {
    // rough ctx estimate via block-table width (host-side, no sync)
    const int64_t approx_L = (int64_t)max_blocks * kBS;
    if (approx_L > 3 * q.size(0)) return false;
  }
  // workgroup width: 8 waves (256 q rows/WG, best arithmetic intensity)
  // unless that would underfill the 256-CU chip — small prefills then
  // run the 4-wave form, whose 2x workgroup count fills more CUs
  // (measured: +11% at Tq=1024, -6% at Tq=8192 — same-box A/B).
  // KUBEAI_V2_WAVES forces either.
  static const int forced_waves = []() {
    const char* e = getenv("KUBEAI_V2_WAVES");
    return e ? atoi(e) : 0;
  }();
  int nw = 8;
  if (G <= 4) {
    if (forced_waves == 4 || forced_waves == 8) {
      nw = forced_waves;
    } else {
      const int qrows8 = (8 / G) * kQB;
      const int64_t wgs8 =
          (int64_t)B * n_kv * ((q.size(0) + qrows8 - 1) / qrows8);
      if (wgs8 < 256) nw = 4;
    }
  }
  // max q chunk rows: nw/G waves * 32
  const int qrows = (nw / G) * kQB;
  int max_qlen = 0;
  {
    // host copy of query_start_loc is cheap (B+1 ints, pinned path); the
    // engine already keeps it on device — derive grid.z from q length
    // bound instead: Tq <= q.size(0)
    max_qlen = q.size(0);
  }
  const int zdim = (max_qlen + qrows - 1) / qrows;
  static const int dbg_mode = []() {
    const char* e = getenv("KUBEAI_V2_DBG");
    return e ? atoi(e) : 0;
  }();
  dim3 grid(B, n_kv, zdim), block(nw * 64);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  const bool fp8_cache = k_cache.scalar_type() == torch::kFloat8_e5m2;
  static const int occ = []() {
    const char* e = getenv("KUBEAI_V2_OCC");
    return e ? atoi(e) : 2;
  }();
#define LAUNCH_V2_CT_DW(GG, CT, D, W)                                     \
  hipLaunchKernelGGL((paged_prefill_v2_kernel<GG, CT, D, W, 8>), grid,    \
                     block,                                               \
                     0, stream, (ushort*)out.data_ptr(),                  \
                     (const ushort*)q.data_ptr(),                         \
                     (const CT*)k_cache.data_ptr(),                       \
                     (const CT*)v_cache.data_ptr(),                       \
                     block_tables.data_ptr<int32_t>(),                    \
                     query_start_loc.data_ptr<int32_t>(),                 \
                     seq_lens.data_ptr<int32_t>(), (float)scale, n_kv,    \
                     max_blocks, q.stride(0))
#define LAUNCH_V2_NW4(GG, CT)                                             \
  hipLaunchKernelGGL((paged_prefill_v2_kernel<GG, CT, 0, 2, 4>), grid,    \
                     block,                                               \
                     0, stream, (ushort*)out.data_ptr(),                  \
                     (const ushort*)q.data_ptr(),                         \
                     (const CT*)k_cache.data_ptr(),                       \
                     (const CT*)v_cache.data_ptr(),                       \
                     block_tables.data_ptr<int32_t>(),                    \
                     query_start_loc.data_ptr<int32_t>(),                 \
                     seq_lens.data_ptr<int32_t>(), (float)scale, n_kv,    \
                     max_blocks, q.stride(0))
#define LAUNCH_V2_CT(GG, CT)                                              \
  do {                                                                    \
    switch (dbg_mode & 3) {                                               \
      case 1: LAUNCH_V2_CT_DW(GG, CT, 1, 2); break;                       \
      case 2: LAUNCH_V2_CT_DW(GG, CT, 2, 2); break;                       \
      case 3: LAUNCH_V2_CT_DW(GG, CT, 3, 2); break;                       \
      default:                                                            \
        if (occ == 4) LAUNCH_V2_CT_DW(GG, CT, 0, 4);                      \
        else if (nw == 4) LAUNCH_V2_NW4(GG, CT);                          \
        else LAUNCH_V2_CT_DW(GG, CT, 0, 2);                               \
        break;                                                            \
    }                                                                     \
  } while (0)
#define LAUNCH_V2(GG)                                                     \
  do {                                                                    \
    if (fp8_cache) {                                                      \
      LAUNCH_V2_CT(GG, unsigned char);                                    \
    } else {                                                              \
      LAUNCH_V2_CT(GG, ushort);                                           \
    }                                                                     \
  } while (0)
  switch (G) {
    case 1: LAUNCH_V2(1); break;
    case 2: LAUNCH_V2(2); break;
    case 4: LAUNCH_V2(4); break;
    case 8: LAUNCH_V2(8); break;
    default: return false;
  }
#undef LAUNCH_V2
#undef LAUNCH_V2_CT
#undef LAUNCH_V2_NW4
#undef LAUNCH_V2_CT_DW
  HIP_CHECK_KERNEL();
  return true;
}
