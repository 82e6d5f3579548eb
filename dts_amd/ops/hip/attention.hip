// Paged attention kernels for MI355X (gfx950, CDNA4).
//
// Decode (one query token per sequence): memory-bound streaming of the
// paged KV; one 4-wave workgroup per (sequence, kv-head), the G q-heads of
// the GQA group computed together so every K/V byte is read once for G
// dot products. Waves split the KV pages round-robin and combine their
// online-softmax partials through LDS (flash-decoding style).
//
// Prefill (chunked, causal, attends cached prefix + own chunk): MFMA
// (v_mfma_f32_16x16x32_bf16) flash-style kernel; one 4-wave workgroup per
// (sequence, 64-row q-tile, kv-head) where a row is a (position, q-head)
// pair; K/V pages staged in LDS (+8-element row padding against bank
// conflicts — the XOR-swizzle alternative measured NEGATIVE here, see the
// SWZ template note below), online softmax with per-row running max/sum.
// Remaining ladder (tr_b16 V reads, async-stage split, 8-phase pipeline)
// is round-2 work; prefill attention is ~8% of the search wall.
//
// Semantics match dts_amd/ops/torch_ref.py attn_*_paged.

#include "common.h"
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <unordered_map>
#include <array>
#include <cstdlib>

using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

// ---------------------------------------------------------------------------
// Decode
// ---------------------------------------------------------------------------

// G = q-heads per kv-head, D = head dim. Block size BS = 16 tokens.
// Split-KV (flash-decoding): grid.z splits each sequence's pages into S
// contiguous ranges; each (seq, kvh, split) workgroup computes partial
// (m, l, o) into the fp32 workspace, combined by attn_decode_combine.
// S is FIXED per launch so decode steps stay hipGraph-capturable with
// varying kv_lens; splits past a short sequence's pages exit immediately.
template <int G, int D>
__global__ void __launch_bounds__(256)
attn_decode_kernel(float* __restrict__ ws_m,  // [B, Hkv, S, G]
                   float* __restrict__ ws_l,  // [B, Hkv, S, G]
                   float* __restrict__ ws_o,  // [B, Hkv, S, G, D]
                   const short* __restrict__ q,      // [B, Hq, D]
                   const short* __restrict__ kcache, // [N, Hkv, BS, D]
                   const short* __restrict__ vcache,
                   const int* __restrict__ block_tables, // [B, max_blocks]
                   const int* __restrict__ kv_lens,      // [B]
                   int max_blocks, int Hkv, float scale, long q_tstride) {
  constexpr int BS = 16;
  constexpr int NWAVE = 4;
  constexpr int LPK = 4;            // lanes per key
  constexpr int DPL = D / LPK;      // dims per lane in K phase
  constexpr int DPV = D / WAVE;     // output dims per lane (2 @128, 1 @64)
  const int seq = blockIdx.x;
  const int kvh = blockIdx.y;
  const int split = blockIdx.z;
  const int S = gridDim.z;
  const int Hq = Hkv * G;
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int kv_len = kv_lens[seq];
  const int n_pages = (kv_len + BS - 1) / BS;
  const int pages_per_split = (n_pages + S - 1) / S;
  const int page_lo = split * pages_per_split;
  const int page_hi = min(n_pages, page_lo + pages_per_split);
  const long ws_base = (((long)seq * Hkv + kvh) * S + split) * G;
  if (page_lo >= page_hi) {
    // empty split: publish -inf partials so the combine skips it
    for (int g = threadIdx.x; g < G; g += blockDim.x) {
      ws_m[ws_base + g] = -1e30f;
      ws_l[ws_base + g] = 0.f;
    }
    return;
  }

  // LDS: per-wave score buffer + combine buffers
  __shared__ float s_scores[NWAVE][G][BS];
  __shared__ float s_m[NWAVE][G], s_l[NWAVE][G];
  __shared__ float s_o[NWAVE][G][D];

  // q fragment: lane covers key-group dims [c*DPL, (c+1)*DPL) with c=lane&3.
  // Kept PACKED (bf16 pairs in u32) to stay under the VGPR occupancy cliff
  // (G=4: 64 regs of q instead of 128 fp32).
  const int c = lane & (LPK - 1);
  const int key_of_lane = lane / LPK; // 0..15
  int qp_[G][DPL / 2];
#pragma unroll
  for (int g = 0; g < G; ++g) {
    const short* qp = q + (long)seq * q_tstride + (long)(kvh * G + g) * D + c * DPL;
#pragma unroll
    for (int i = 0; i < DPL / 2; ++i) qp_[g][i] = ((const int*)qp)[i];
  }

  float m[G], lsum[G], o[G][DPV]; // lane owns dims [lane*DPV, lane*DPV+DPV)
#pragma unroll
  for (int g = 0; g < G; ++g) {
    m[g] = -1e30f;
    lsum[g] = 0.f;
#pragma unroll
    for (int d = 0; d < DPV; ++d) o[g][d] = 0.f;
  }

  const int* bt = block_tables + (long)seq * max_blocks;

  for (int page = page_lo + wave; page < page_hi; page += NWAVE) {
    const long blk = bt[page];
    const short* kbase =
        kcache + ((blk * Hkv + kvh) * BS) * (long)D;
    const short* vbase =
        vcache + ((blk * Hkv + kvh) * BS) * (long)D;
    const int valid = min(BS, kv_len - page * BS);

    // ---- K phase: scores for the 16 keys of this page
    float partial[G];
#pragma unroll
    for (int g = 0; g < G; ++g) partial[g] = 0.f;
    if (key_of_lane < valid) {
      const short* kp = kbase + key_of_lane * D + c * DPL;
#pragma unroll
      for (int i = 0; i < DPL / 2; ++i) {
        int kw = ((const int*)kp)[i];
        float k0 = bf2f((short)(kw & 0xffff));
        float k1 = bf2f((short)((kw >> 16) & 0xffff));
#pragma unroll
        for (int g = 0; g < G; ++g) {
          float q0 = bf2f((short)(qp_[g][i] & 0xffff));
          float q1 = bf2f((short)((qp_[g][i] >> 16) & 0xffff));
          partial[g] += q0 * k0 + q1 * k1;
        }
      }
    }
    // reduce over the 4 lanes of the key group
#pragma unroll
    for (int g = 0; g < G; ++g) {
      partial[g] += __shfl_xor(partial[g], 1, 64);
      partial[g] += __shfl_xor(partial[g], 2, 64);
      if (c == 0)
        s_scores[wave][g][key_of_lane] =
            (key_of_lane < valid) ? partial[g] * scale : -1e30f;
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_wave_barrier();

    // ---- batch the page's V words up front: 16 independent loads in
    // flight at once (the per-(g,j) load placement serialized an L2/HBM
    // round trip into the online-softmax chain — ~5 us/page at long kv)
    int vw[BS];
#pragma unroll
    for (int j = 0; j < BS; ++j) {
      if (j < valid) {
        if constexpr (DPV == 2)
          vw[j] = *(const int*)(vbase + j * D + 2 * lane);
        else
          vw[j] = (int)*(const unsigned short*)(vbase + j * D + lane);
      } else {
        vw[j] = 0;
      }
    }

    // ---- softmax update + V accumulate; lane owns output dims {2l, 2l+1}
#pragma unroll
    for (int g = 0; g < G; ++g) {
      float tmax = -1e30f;
#pragma unroll
      for (int j = 0; j < BS; ++j) tmax = fmaxf(tmax, s_scores[wave][g][j]);
      float m_new = fmaxf(m[g], tmax);
      if (m_new > -1e30f) {
        float alpha = __expf(m[g] - m_new);
#pragma unroll
        for (int d = 0; d < DPV; ++d) o[g][d] *= alpha;
        lsum[g] *= alpha;
        m[g] = m_new;
        float p[BS];
        float psum = 0.f;
#pragma unroll
        for (int j = 0; j < BS; ++j) {
          p[j] = (j < valid) ? __expf(s_scores[wave][g][j] - m_new) : 0.f;
          psum += p[j];
        }
        lsum[g] += psum;
#pragma unroll
        for (int j = 0; j < BS; ++j) {
          o[g][0] += p[j] * bf2f((short)(vw[j] & 0xffff));
          if constexpr (DPV == 2)
            o[g][1] += p[j] * bf2f((short)((vw[j] >> 16) & 0xffff));
        }
      }
    }
    __builtin_amdgcn_wave_barrier();
  }

  // ---- cross-wave combine via LDS, then publish split partials
#pragma unroll
  for (int g = 0; g < G; ++g) {
    if (lane == 0) {
      s_m[wave][g] = m[g];
      s_l[wave][g] = lsum[g];
    }
#pragma unroll
    for (int d = 0; d < DPV; ++d) s_o[wave][g][lane * DPV + d] = o[g][d];
  }
  __syncthreads();

  const int tid = threadIdx.x;
  for (int gd = tid; gd < G * (D / 2); gd += blockDim.x) {
    const int g = gd / (D / 2);
    const int d0 = (gd % (D / 2)) * 2;
    float m_star = -1e30f;
#pragma unroll
    for (int w = 0; w < NWAVE; ++w) m_star = fmaxf(m_star, s_m[w][g]);
    float l_star = 0.f, acc0 = 0.f, acc1 = 0.f;
#pragma unroll
    for (int w = 0; w < NWAVE; ++w) {
      float f = (s_m[w][g] > -1e30f) ? __expf(s_m[w][g] - m_star) : 0.f;
      l_star += s_l[w][g] * f;
      acc0 += s_o[w][g][d0] * f;
      acc1 += s_o[w][g][d0 + 1] * f;
    }
    if (d0 == 0) {
      ws_m[ws_base + g] = m_star;
      ws_l[ws_base + g] = l_star;
    }
    float* wo = ws_o + (ws_base + g) * D;
    wo[d0] = acc0;
    wo[d0 + 1] = acc1;
  }
}  // (D/2 pairs: valid for D=64 too — s_o is [.][.][D])

// Combine the S split partials into the final bf16 output.
template <int D>
__global__ void __launch_bounds__(256)
attn_decode_combine(short* __restrict__ out,        // [B, Hq, D]
                    const float* __restrict__ ws_m, // [B, Hkv, S, G]
                    const float* __restrict__ ws_l,
                    const float* __restrict__ ws_o, // [B, Hkv, S, G, D]
                    int S, int G, int Hkv) {
  const int seq = blockIdx.x;
  const int hq = blockIdx.y;  // 0..Hq-1
  const int kvh = hq / G;
  const int g = hq % G;
  const int Hq = Hkv * G;
  const long base0 = (((long)seq * Hkv + kvh) * S) * G + g;

  __shared__ float sh_f[64];  // per-split weight
  __shared__ float sh_linv[1];
  // pass 1: global max + weighted l (thread 0 does the tiny reduction)
  if (threadIdx.x == 0) {
    float m_star = -1e30f;
    for (int s2 = 0; s2 < S; ++s2)
      m_star = fmaxf(m_star, ws_m[base0 + (long)s2 * G]);
    float l_star = 0.f;
    for (int s2 = 0; s2 < S; ++s2) {
      float mm = ws_m[base0 + (long)s2 * G];
      float f = (mm > -1e30f) ? __expf(mm - m_star) : 0.f;
      sh_f[s2] = f;
      l_star += ws_l[base0 + (long)s2 * G] * f;
    }
    sh_linv[0] = (l_star > 0.f) ? 1.f / l_star : 0.f;
  }
  __syncthreads();
  const float linv = sh_linv[0];
  for (int d = threadIdx.x; d < D; d += blockDim.x) {
    float acc = 0.f;
    for (int s2 = 0; s2 < S; ++s2) {
      float f = sh_f[s2];
      if (f > 0.f) acc += ws_o[(base0 + (long)s2 * G) * D + d] * f;
    }
    out[(((long)seq * Hq) + hq) * D + d] = f2bf(acc * linv);
  }
}

// ---------------------------------------------------------------------------
// Prefill (MFMA) v2 — KSTEP=64, reg-staged K/V pipeline, transposed V LDS
//
// v1 (below) measured 137/180/220 TF at N=2k/4k/8k; its three structural
// costs, per the guide's attention ladder: (1) KSTEP=32 pays two full
// barriers per 32 keys, (2) the K/V tile load is synchronous — ~500
// cycles of HBM latency exposed per tile, (3) the P·V B-fragment reads V
// columns as 64 scalar ds_read_u16 per lane per tile. v2: 64-key tiles
// halve the barrier rate; the NEXT tile's K/V global loads issue into
// registers before the current tile's compute (T14 issue-early /
// write-late) so HBM latency hides under the MFMAs; V stores into a
// transposed [D][KSTEP] LDS image (row stride 68 elements: 8-byte
// aligned for b64 reads, 4-distinct-bank store pattern) so each P·V
// B-fragment is two contiguous ds_read_b64 instead of eight u16 gathers.
// ---------------------------------------------------------------------------

template <int G, int D, int KSTEP>
__global__ void __launch_bounds__(256)
attn_prefill_kernel_v2(short* __restrict__ out,      // [Tq, Hq, D]
                       const short* __restrict__ q,  // [Tq, Hq, D]
                       const int* __restrict__ cu_q, // [P+1]
                       const long* __restrict__ q_pos, // [Tq]
                       const short* __restrict__ kcache,
                       const short* __restrict__ vcache,
                       const int* __restrict__ block_tables,
                       const int* __restrict__ kv_lens, int max_blocks,
                       int Hkv, float scale, long q_tstride) {
  constexpr int BS = 16;
  constexpr int KCHUNKS = D / 32;     // mfma k-chunks per QK^T
  constexpr int CTILES = D / 16;      // 16-col output tiles
  constexpr int LDK = D + 8;          // K row stride (elements)
  constexpr int LDV = KSTEP + 4;      // transposed-V row stride: 68
  constexpr int ROWS = 64;
  const int POS_PER_WG = ROWS / G;

  const int seq = blockIdx.y;
  const int kvh = blockIdx.z;
  const int Hq = Hkv * G;
  const int q_start = cu_q[seq];
  const int q_len = cu_q[seq + 1] - q_start;
  const int tile = blockIdx.x;
  if (tile * POS_PER_WG >= q_len) return;

  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int kv_len = kv_lens[seq];
  const int* bt = block_tables + (long)seq * max_blocks;

  __shared__ short k_lds[KSTEP][LDK];
  __shared__ short v_lds_t[D][LDV];   // transposed: [dim][key]
  __shared__ float s_scores[4][16][KSTEP];
  __shared__ float s_alpha[4][16];
  __shared__ float s_rowl[4][16];
  __shared__ short p_lds[4][16][KSTEP];

  // Q A-fragments (persistent)
  short a_frag[KCHUNKS][8];
  {
    const int r = lane & 15;
    const int row_global = wave * 16 + r;
    const int pos_local = tile * POS_PER_WG + row_global / G;
    const int head = row_global % G;
    const bool valid_row = pos_local < q_len;
    const int tok = q_start + (valid_row ? pos_local : 0);
    const short* qp = q + (long)tok * q_tstride + (long)(kvh * G + head) * D;
#pragma unroll
    for (int kc = 0; kc < KCHUNKS; ++kc) {
      const int kbase = kc * 32 + (lane >> 4) * 8;
      bf16x8 v8 = *(const bf16x8*)(qp + kbase);
#pragma unroll
      for (int i = 0; i < 8; ++i)
        a_frag[kc][i] = valid_row ? v8[i] : (short)0;
    }
  }
  constexpr int SMW = KSTEP / 4;  // keys per softmax lane
  const int sm_row = lane / 4;
  const int sm_sub = lane & 3;
  const long sm_pos = tile * POS_PER_WG + (wave * 16 + sm_row) / G;
  const bool sm_valid = sm_pos < q_len;
  const long sm_abs_pos = sm_valid ? q_pos[q_start + sm_pos] : -1;

  f32x4 o_acc[CTILES];
#pragma unroll
  for (int ct = 0; ct < CTILES; ++ct) o_acc[ct] = {0.f, 0.f, 0.f, 0.f};
  float run_m = -1e30f, run_l = 0.f;

  const int last_local_pos = min(q_len, tile * POS_PER_WG + POS_PER_WG) - 1;
  const long last_abs_pos = q_pos[q_start + last_local_pos];
  const int kv_hi = min((long)kv_len, last_abs_pos + 1);

  // ---- reg-staged tile loads: each thread owns 4 (key, d0) chunks of 8
  // bf16 for K and the same for V (KSTEP*D / 256 threads / 8 = 4)
  constexpr int CHUNKS_PT = KSTEP * D / (256 * 8);
  bf16x8 k_stage[CHUNKS_PT], v_stage[CHUNKS_PT];
  int st_key[CHUNKS_PT], st_d0[CHUNKS_PT];

  auto issue_tile = [&](int kv_base) {
#pragma unroll
    for (int c = 0; c < CHUNKS_PT; ++c) {
      const int base = (threadIdx.x + c * 256) * 8;
      const int key = base / D;
      const int d0 = base % D;
      st_key[c] = key;
      st_d0[c] = d0;
      const int kglob = kv_base + key;
      if (kglob < kv_len) {
        const long blk = bt[kglob / BS];
        const long off = ((blk * Hkv + kvh) * BS + kglob % BS) * (long)D + d0;
        k_stage[c] = *(const bf16x8*)(kcache + off);
        v_stage[c] = *(const bf16x8*)(vcache + off);
      } else {
        const bf16x8 z = {0, 0, 0, 0, 0, 0, 0, 0};
        k_stage[c] = z;
        v_stage[c] = z;
      }
    }
  };
  auto write_tile = [&]() {
#pragma unroll
    for (int c = 0; c < CHUNKS_PT; ++c) {
      *(bf16x8*)(&k_lds[st_key[c]][st_d0[c]]) = k_stage[c];
#pragma unroll
      for (int i = 0; i < 8; ++i)
        v_lds_t[st_d0[c] + i][st_key[c]] = v_stage[c][i];
    }
  };

  issue_tile(0);
  write_tile();
  __syncthreads();

  for (int kv_base = 0; kv_base < kv_hi; kv_base += KSTEP) {
    const int kv_next = kv_base + KSTEP;
    // T14 issue-early: next tile's HBM loads start before this tile's
    // compute; the LDS write happens after the barrier below
    if (kv_next < kv_hi) issue_tile(kv_next);

    // ---- QK^T: four 16x16 score tiles over KCHUNKS k-chunks
#pragma unroll
    for (int st = 0; st < KSTEP / 16; ++st) {
      f32x4 acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kc = 0; kc < KCHUNKS; ++kc) {
        const int key = st * 16 + (lane & 15);
        const int d0 = kc * 32 + (lane >> 4) * 8;
        bf16x8 b8 = *(bf16x8*)(&k_lds[key][d0]);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            *(bf16x8*)a_frag[kc], b8, acc, 0, 0, 0);
      }
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int row = (lane >> 4) * 4 + i;
        s_scores[wave][row][st * 16 + (lane & 15)] = acc[i] * scale;
      }
    }
    __builtin_amdgcn_s_waitcnt(0);
    __builtin_amdgcn_wave_barrier();

    // ---- online softmax: 4 lanes per row, each scans 16 keys
    {
      float tmax = -1e30f;
      float sc[SMW];
#pragma unroll
      for (int jj = 0; jj < SMW; ++jj) {
        const int j = sm_sub * SMW + jj;
        const long key_abs = kv_base + j;
        float s = s_scores[wave][sm_row][j];
        const bool ok =
            sm_valid && key_abs <= sm_abs_pos && key_abs < (long)kv_len;
        sc[jj] = ok ? s : -1e30f;
        tmax = fmaxf(tmax, sc[jj]);
      }
      tmax = fmaxf(tmax, __shfl_xor(tmax, 1, 64));
      tmax = fmaxf(tmax, __shfl_xor(tmax, 2, 64));
      float m_new = fmaxf(run_m, tmax);
      float alpha, rowsum = 0.f;
      if (m_new > -1e30f) {
        alpha = (run_m > -1e30f) ? __expf(run_m - m_new) : 0.f;
#pragma unroll
        for (int jj = 0; jj < SMW; ++jj) {
          float p = (sc[jj] > -1e30f) ? __expf(sc[jj] - m_new) : 0.f;
          p_lds[wave][sm_row][sm_sub * SMW + jj] = f2bf(p);
          rowsum += p;
        }
      } else {
        alpha = 1.f;
#pragma unroll
        for (int jj = 0; jj < SMW; ++jj)
          p_lds[wave][sm_row][sm_sub * SMW + jj] = 0;
      }
      rowsum += __shfl_xor(rowsum, 1, 64);
      rowsum += __shfl_xor(rowsum, 2, 64);
      run_l = run_l * alpha + rowsum;
      run_m = m_new;
      if (sm_sub == 0) {
        s_alpha[wave][sm_row] = alpha;
        s_rowl[wave][sm_row] = run_l;
      }
    }
    __builtin_amdgcn_s_waitcnt(0);
    __builtin_amdgcn_wave_barrier();

    // ---- rescale O and accumulate P*V (two k=32 chunks over 64 keys)
    {
      float al[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) al[i] = s_alpha[wave][(lane >> 4) * 4 + i];
      constexpr int HCH = KSTEP / 32;
      bf16x8 pa[HCH];
      {
        const int row = lane & 15;
#pragma unroll
        for (int h = 0; h < HCH; ++h) {
          const int k0 = h * 32 + (lane >> 4) * 8;
#pragma unroll
          for (int i = 0; i < 8; ++i) pa[h][i] = p_lds[wave][row][k0 + i];
        }
      }
#pragma unroll
      for (int ct = 0; ct < CTILES; ++ct) {
#pragma unroll
        for (int i = 0; i < 4; ++i) o_acc[ct][i] *= al[i];
        const int dim = ct * 16 + (lane & 15);
#pragma unroll
        for (int h = 0; h < HCH; ++h) {
          // V B-frag: 8 consecutive keys of one dim = 2x ds_read_b64
          // (the 68-element row stride is 8-byte aligned, not 16)
          const int kk0 = h * 32 + (lane >> 4) * 8;
          bf16x8 vb;
          ((unsigned long long*)&vb)[0] =
              *(const unsigned long long*)(&v_lds_t[dim][kk0]);
          ((unsigned long long*)&vb)[1] =
              *(const unsigned long long*)(&v_lds_t[dim][kk0 + 4]);
          o_acc[ct] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              pa[h], vb, o_acc[ct], 0, 0, 0);
        }
      }
    }
    __syncthreads();  // everyone done READING this tile's LDS
    if (kv_next < kv_hi) {
      write_tile();   // T14 write-late: stage the next tile
      __syncthreads();
    }
  }

  // ---- epilogue: normalize + store
  {
    float invl[4];
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      float l = s_rowl[wave][(lane >> 4) * 4 + i];
      invl[i] = (l > 0.f) ? 1.f / l : 0.f;
    }
#pragma unroll
    for (int ct = 0; ct < CTILES; ++ct) {
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int row = (lane >> 4) * 4 + i;
        const int row_global = wave * 16 + row;
        const int pos_local = tile * POS_PER_WG + row_global / G;
        if (pos_local >= q_len) continue;
        const int head = row_global % G;
        const int tok = q_start + pos_local;
        const int dim = ct * 16 + (lane & 15);
        out[((long)tok * Hq + kvh * G + head) * D + dim] =
            f2bf(o_acc[ct][i] * invl[i]);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Prefill (MFMA) v5 — swapped QK^T + in-register softmax (guide T12)
//
// PMC decomposition of v1 (profiles/pmc_prefill_r2.txt): 43% of wave
// cycles PARKED (s_waitcnt/barrier), 24% issue-stalled, 33% active,
// 40% LDS bank-conflict overhead, MFMA util ~4%. The parked+active time
// is dominated by the softmax round trip: scores C-frag -> LDS ->
// per-row scan -> P -> LDS -> P A-frag, with waitcnt(0)+wave_barrier
// fences around each hop. v5 computes the TRANSPOSED score tile
// C'[key, q] = mfma(A = K-frag, B = Q-frag) — the persistent Q
// registers already have the B layout — so each lane holds its q
// column's scores in registers; max/sum reduce with two shfl_xor ops,
// P converts to the P·V A-fragment with 4 packs + 4 shfls, and the
// scores/p LDS buffers AND their fences disappear (LDS drops 24 KB ->
// more workgroups per CU).
// ---------------------------------------------------------------------------

template <int G, int D, int KSTEP>
__global__ void __launch_bounds__(256)
attn_prefill_kernel_v5(short* __restrict__ out,      // [Tq, Hq, D]
                       const short* __restrict__ q,  // [Tq, Hq, D]
                       const int* __restrict__ cu_q, // [P+1]
                       const long* __restrict__ q_pos, // [Tq]
                       const short* __restrict__ kcache,
                       const short* __restrict__ vcache,
                       const int* __restrict__ block_tables,
                       const int* __restrict__ kv_lens, int max_blocks,
                       int Hkv, float scale, long q_tstride) {
  constexpr int BS = 16;
  constexpr int KCHUNKS = D / 32;
  constexpr int CTILES = D / 16;
  constexpr int LDS_PAD = 8;
  constexpr int LDK = D + LDS_PAD;
  constexpr int ROWS = 64;
  const int POS_PER_WG = ROWS / G;

  const int seq = blockIdx.y;
  const int kvh = blockIdx.z;
  const int Hq = Hkv * G;
  const int q_start = cu_q[seq];
  const int q_len = cu_q[seq + 1] - q_start;
  const int tile = blockIdx.x;
  if (tile * POS_PER_WG >= q_len) return;

  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int kv_len = kv_lens[seq];
  const int* bt = block_tables + (long)seq * max_blocks;

  __shared__ short k_lds[KSTEP][LDK];
  __shared__ short v_lds[KSTEP][LDK];

  // Q B-fragments (persistent; layout identical to v1's A-frag)
  short q_frag[KCHUNKS][8];
  {
    const int r = lane & 15;
    const int row_global = wave * 16 + r;
    const int pos_local = tile * POS_PER_WG + row_global / G;
    const int head = row_global % G;
    const bool valid_row = pos_local < q_len;
    const int tok = q_start + (valid_row ? pos_local : 0);
    const short* qp = q + (long)tok * q_tstride + (long)(kvh * G + head) * D;
#pragma unroll
    for (int kc = 0; kc < KCHUNKS; ++kc) {
      const int kbase = kc * 32 + (lane >> 4) * 8;
      bf16x8 v8 = *(const bf16x8*)(qp + kbase);
#pragma unroll
      for (int i = 0; i < 8; ++i)
        q_frag[kc][i] = valid_row ? v8[i] : (short)0;
    }
  }
  // this lane's softmax q (col of the swapped score tile)
  const int my_q = lane & 15;
  const int my_row_global = wave * 16 + my_q;
  const long my_pos_local = tile * POS_PER_WG + my_row_global / G;
  const bool my_valid = my_pos_local < q_len;
  const long my_abs_pos = my_valid ? q_pos[q_start + my_pos_local] : -1;

  f32x4 o_acc[CTILES];
#pragma unroll
  for (int ct = 0; ct < CTILES; ++ct) o_acc[ct] = {0.f, 0.f, 0.f, 0.f};
  float run_m = -1e30f, run_l = 0.f;  // for q = my_q (replicated x4 lanes)

  const int last_local_pos = min(q_len, tile * POS_PER_WG + POS_PER_WG) - 1;
  const long last_abs_pos = q_pos[q_start + last_local_pos];
  const int kv_hi = min((long)kv_len, last_abs_pos + 1);

  for (int kv_base = 0; kv_base < kv_hi; kv_base += KSTEP) {
    // ---- cooperative K/V tile load (shared by all 4 waves)
    {
      const int elems = KSTEP * D;
      for (int base = threadIdx.x * 8; base < elems; base += 256 * 8) {
        const int key = base / D;
        const int d0 = base % D;
        const int kglob = kv_base + key;
        if (kglob < kv_len) {
          const long blk = bt[kglob / BS];
          const long off =
              ((blk * Hkv + kvh) * BS + kglob % BS) * (long)D + d0;
          *(bf16x8*)(&k_lds[key][d0]) = *(const bf16x8*)(kcache + off);
          *(bf16x8*)(&v_lds[key][d0]) = *(const bf16x8*)(vcache + off);
        } else {
          const bf16x8 z = {0, 0, 0, 0, 0, 0, 0, 0};
          *(bf16x8*)(&k_lds[key][d0]) = z;
          *(bf16x8*)(&v_lds[key][d0]) = z;
        }
      }
    }
    __syncthreads();

    // ---- swapped QK^T: C'[key, q] tiles, KSTEP/16 of them
    constexpr int STILES = KSTEP / 16;
    f32x4 sc[STILES];
#pragma unroll
    for (int st = 0; st < STILES; ++st) {
      f32x4 acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kc = 0; kc < KCHUNKS; ++kc) {
        // A-frag = K rows: row = st*16 + (lane&15), k-dim = d chunk
        const int key = st * 16 + (lane & 15);
        const int d0 = kc * 32 + (lane >> 4) * 8;
        bf16x8 k8 = *(bf16x8*)(&k_lds[key][d0]);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            k8, *(bf16x8*)q_frag[kc], acc, 0, 0, 0);
      }
      sc[st] = acc;  // rows: key = st*16 + (lane>>4)*4 + i; col: my_q
    }

    // ---- in-register online softmax for column my_q
    float p[STILES][4];
    float tmax = -1e30f;
#pragma unroll
    for (int st = 0; st < STILES; ++st)
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const long key_abs = kv_base + st * 16 + (lane >> 4) * 4 + i;
        const bool ok =
            my_valid && key_abs <= my_abs_pos && key_abs < (long)kv_len;
        p[st][i] = ok ? sc[st][i] * scale : -1e30f;
        tmax = fmaxf(tmax, p[st][i]);
      }
    tmax = fmaxf(tmax, __shfl_xor(tmax, 16, 64));
    tmax = fmaxf(tmax, __shfl_xor(tmax, 32, 64));
    const float m_new = fmaxf(run_m, tmax);
    float alpha;
    float rowsum = 0.f;
    if (m_new > -1e30f) {
      alpha = (run_m > -1e30f) ? __expf(run_m - m_new) : 0.f;
#pragma unroll
      for (int st = 0; st < STILES; ++st)
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          p[st][i] = (p[st][i] > -1e30f) ? __expf(p[st][i] - m_new) : 0.f;
          rowsum += p[st][i];
        }
    } else {
      alpha = 1.f;
#pragma unroll
      for (int st = 0; st < STILES; ++st)
#pragma unroll
        for (int i = 0; i < 4; ++i) p[st][i] = 0.f;
    }
    rowsum += __shfl_xor(rowsum, 16, 64);
    rowsum += __shfl_xor(rowsum, 32, 64);
    run_l = run_l * alpha + rowsum;
    run_m = m_new;

    // ---- P' -> P·V A-fragments in registers: pack adjacent-key bf16
    // pairs, then gather the 8-consecutive-key groups across the 4
    // lane-groups (keys 4*(l>>4)+i live here; target wants 8*(l>>4)+i)
    unsigned int ppack[STILES][2];
#pragma unroll
    for (int st = 0; st < STILES; ++st) {
#pragma unroll
      for (int hpair = 0; hpair < 2; ++hpair) {
        const unsigned int lo =
            (unsigned short)f2bf(p[st][hpair * 2]);
        const unsigned int hi =
            (unsigned short)f2bf(p[st][hpair * 2 + 1]);
        ppack[st][hpair] = lo | (hi << 16);
      }
    }
    // per PV k-chunk h (32 keys): A-frag lane needs keys
    // 8*(lane>>4)..+8 of chunk h — i.e. packs from source groups
    // g0 = 2*(lane>>4), g1 = 2*(lane>>4)+1 of score-tile st = h*2 +
    // (g/ (KSTEP/32))... for KSTEP=32 there are 2 score tiles (32 keys)
    // and ONE PV chunk; for KSTEP=64, 4 tiles and 2 chunks.
    constexpr int HCH = KSTEP / 32;
    bf16x8 pa[HCH];
#pragma unroll
    for (int h = 0; h < HCH; ++h) {
      // global key base for this lane's A-frag: h*32 + 8*(lane>>4).
      // Keys kb..kb+3 live in score tile st_a / lane-group ga, keys
      // kb+4..kb+7 in st_b / gb. A shfl OPERAND must be indexed
      // uniformly across lanes (each lane evaluates its own
      // expression), so gather every tile with uniform indices and
      // select by this lane's st_a/st_b.
      const int kb = 8 * (lane >> 4);     // within the 32-key chunk
      const int st_a = h * 2 + kb / 16;   // score tile holding kb..kb+3
      const int ga = (kb % 16) / 4;       // lane-group that owns them
      const int st_b = h * 2 + (kb + 4) / 16;
      const int gb = ((kb + 4) % 16) / 4;
      unsigned int paw[4];
#pragma unroll
      for (int pr = 0; pr < 2; ++pr) {
        unsigned int va = 0, vb2 = 0;
#pragma unroll
        for (int st = 0; st < STILES; ++st) {
          const unsigned int ta =
              (unsigned int)__shfl((int)ppack[st][pr], (ga << 4) | my_q, 64);
          const unsigned int tb =
              (unsigned int)__shfl((int)ppack[st][pr], (gb << 4) | my_q, 64);
          if (st == st_a) va = ta;
          if (st == st_b) vb2 = tb;
        }
        paw[pr] = va;
        paw[2 + pr] = vb2;
      }
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        pa[h][2 * j] = (short)(paw[j] & 0xffff);
        pa[h][2 * j + 1] = (short)(paw[j] >> 16);
      }
    }
    // rescale O by alpha of row q' = (lane>>4)*4+i (held by lane q')
#pragma unroll
    for (int ct = 0; ct < CTILES; ++ct) {
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const float al = __shfl(alpha, (lane >> 4) * 4 + i, 64);
        o_acc[ct][i] *= al;
      }
    }
#pragma unroll
    for (int ct = 0; ct < CTILES; ++ct) {
      const int dim = ct * 16 + (lane & 15);
#pragma unroll
      for (int h = 0; h < HCH; ++h) {
        bf16x8 vb;
        const int kk0 = h * 32 + (lane >> 4) * 8;
#pragma unroll
        for (int i = 0; i < 8; ++i) vb[i] = v_lds[kk0 + i][dim];
        o_acc[ct] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            pa[h], vb, o_acc[ct], 0, 0, 0);
      }
    }
    __syncthreads();
  }

  // ---- epilogue: normalize + store (invl of row q' via shfl)
  {
#pragma unroll
    for (int ct = 0; ct < CTILES; ++ct) {
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int row = (lane >> 4) * 4 + i;
        const float l = __shfl(run_l, row, 64);
        const float invl = (l > 0.f) ? 1.f / l : 0.f;
        const int row_global = wave * 16 + row;
        const int pos_local = tile * POS_PER_WG + row_global / G;
        if (pos_local >= q_len) continue;
        const int head = row_global % G;
        const int tok = q_start + pos_local;
        const int dim = ct * 16 + (lane & 15);
        out[((long)tok * Hq + kvh * G + head) * D + dim] =
            f2bf(o_acc[ct][i] * invl);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Prefill (MFMA) v1 — KSTEP=32 reference implementation
// ---------------------------------------------------------------------------

// Rows per workgroup: 64 = (64/G) positions x G heads. Each wave owns 16
// rows. KV step = 32 keys (2 pages). D in {64, 128}.
// SWZ: XOR-swizzle the K/V LDS images (guide T2): element column block
// col' = col ^ ((key & 7) << 3), applied identically at the cooperative
// store and every read — spreads the b128 B-frag lane groups across bank
// slots. A/B-selectable per call (scripts/prefill_probe.py).
template <int G, int D, bool SWZ>
__global__ void __launch_bounds__(256)
attn_prefill_kernel(short* __restrict__ out,      // [Tq, Hq, D]
                    const short* __restrict__ q,  // [Tq, Hq, D]
                    const int* __restrict__ cu_q, // [P+1]
                    const long* __restrict__ q_pos, // [Tq]
                    const short* __restrict__ kcache,
                    const short* __restrict__ vcache,
                    const int* __restrict__ block_tables,
                    const int* __restrict__ kv_lens, int max_blocks, int Hkv,
                    float scale, long q_tstride) {
  constexpr int BS = 16;
  constexpr int KSTEP = 32;
  constexpr int KCHUNKS = D / 32;     // mfma k-chunks per QK^T
  constexpr int CTILES = D / 16;      // 16-col output tiles
  constexpr int LDS_PAD = 8;          // bf16 elements of row padding
  constexpr int LDK = D + LDS_PAD;    // LDS row stride (elements)
  constexpr int ROWS = 64;            // rows per workgroup
  const int POS_PER_WG = ROWS / G;

  const int seq = blockIdx.y;
  const int kvh = blockIdx.z;
  const int Hq = Hkv * G;
  const int q_start = cu_q[seq];
  const int q_len = cu_q[seq + 1] - q_start;
  const int tile = blockIdx.x;
  if (tile * POS_PER_WG >= q_len) return;

  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int kv_len = kv_lens[seq];
  const int* bt = block_tables + (long)seq * max_blocks;

  __shared__ short k_lds[KSTEP][LDK];
  __shared__ short v_lds[KSTEP][LDK];
  // element-column swizzle within a row (8-element blocks stay contiguous)
  auto swz_col = [](int key, int col) -> int {
    if constexpr (SWZ) return ((col & ~7) ^ ((key & 7) << 3)) | (col & 7);
    else return col;
  };
  __shared__ float s_scores[4][16][KSTEP];
  __shared__ float s_alpha[4][16];
  __shared__ float s_rowl[4][16];
  __shared__ short p_lds[4][16][KSTEP];

  // ---- per-wave row bookkeeping: row_global = wave*16 + r
  // row -> (position, head)
  long my_pos[4];   // absolute position of rows owned by this lane group
  // Q A-fragments: KCHUNKS k-chunks of 32, 8 bf16 each
  short a_frag[KCHUNKS][8];
  {
    const int r = lane & 15;            // A row = lane & 15
    const int row_global = wave * 16 + r;
    const int pos_local = tile * POS_PER_WG + row_global / G;
    const int head = row_global % G;
    const bool valid_row = pos_local < q_len;
    const int tok = q_start + (valid_row ? pos_local : 0);
    const short* qp = q + (long)tok * q_tstride + (long)(kvh * G + head) * D;
#pragma unroll
    for (int kc = 0; kc < KCHUNKS; ++kc) {
      const int kbase = kc * 32 + (lane >> 4) * 8;
      bf16x8 v8 = *(const bf16x8*)(qp + kbase);
#pragma unroll
      for (int i = 0; i < 8; ++i)
        a_frag[kc][i] = valid_row ? v8[i] : (short)0;
    }
  }
  // rows this lane owns in the softmax phase: row = lane / 4 (4 lanes/row)
  const int sm_row = lane / 4;
  const int sm_sub = lane & 3;  // which 8-key span this lane scans
  const long sm_pos =
      tile * POS_PER_WG + (wave * 16 + sm_row) / G;  // local position
  const bool sm_valid = sm_pos < q_len;
  const long sm_abs_pos = sm_valid ? q_pos[q_start + sm_pos] : -1;

  // O accumulator: CTILES column tiles of C-frag f32x4
  f32x4 o_acc[CTILES];
#pragma unroll
  for (int ct = 0; ct < CTILES; ++ct) o_acc[ct] = {0.f, 0.f, 0.f, 0.f};
  float run_m = -1e30f, run_l = 0.f;  // per (lane, sm) bookkeeping below
  // running m/l per row are tracked by the 4-lane row group (same value
  // in all 4 lanes; reduced via shfl)

  // max absolute position in this workgroup bounds the causal KV range
  // (rows of this WG cover positions tile*PPW .. tile*PPW+PPW-1)
  const int last_local_pos = min(q_len, tile * POS_PER_WG + POS_PER_WG) - 1;
  const long last_abs_pos = q_pos[q_start + last_local_pos];
  const int kv_hi = min((long)kv_len, last_abs_pos + 1);

  for (int kv_base = 0; kv_base < kv_hi; kv_base += KSTEP) {
    // ---- cooperative K/V tile load (2 pages), 16B per thread per pass
    {
      const int elems = KSTEP * D;          // 32*128 bf16
      // thread t loads 8 bf16 at flat index t*8 (+ stride)
      for (int base = threadIdx.x * 8; base < elems; base += 256 * 8) {
        const int key = base / D;
        const int d0 = base % D;
        const int page = kv_base / BS + key / BS;
        const int koff = key % BS;
        const int kglob = kv_base + key;
        const int dst = swz_col(key, d0);
        if (kglob < kv_len) {
          const long blk = bt[page];
          const short* kp =
              kcache + ((blk * Hkv + kvh) * BS + koff) * (long)D + d0;
          const short* vp =
              vcache + ((blk * Hkv + kvh) * BS + koff) * (long)D + d0;
          *(bf16x8*)(&k_lds[key][dst]) = *(const bf16x8*)kp;
          *(bf16x8*)(&v_lds[key][dst]) = *(const bf16x8*)vp;
        } else {
          bf16x8 z = {0, 0, 0, 0, 0, 0, 0, 0};
          *(bf16x8*)(&k_lds[key][dst]) = z;
          *(bf16x8*)(&v_lds[key][dst]) = z;
        }
      }
    }
    __syncthreads();

    // ---- QK^T: two 16x16 score tiles over 4 k-chunks
#pragma unroll
    for (int st = 0; st < 2; ++st) {
      f32x4 acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kc = 0; kc < KCHUNKS; ++kc) {
        // B-frag: col = lane&15 -> key = st*16 + (lane&15),
        //         k  = kc*32 + (lane>>4)*8 + i  (8 consecutive dims)
        const int key = st * 16 + (lane & 15);
        const int d0 = kc * 32 + (lane >> 4) * 8;
        bf16x8 b8 = *(const bf16x8*)(&k_lds[key][swz_col(key, d0)]);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            *(bf16x8*)a_frag[kc], b8, acc, 0, 0, 0);
      }
      // scatter C-frag to LDS scores: row=(lane>>4)*4+i, col=lane&15
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int row = (lane >> 4) * 4 + i;
        s_scores[wave][row][st * 16 + (lane & 15)] = acc[i] * scale;
      }
    }
    __builtin_amdgcn_s_waitcnt(0);
    __builtin_amdgcn_wave_barrier();

    // ---- online softmax: 4 lanes per row, each scans 8 keys
    {
      float tmax = -1e30f;
      float sc[8];
#pragma unroll
      for (int jj = 0; jj < 8; ++jj) {
        const int j = sm_sub * 8 + jj;
        const long key_abs = kv_base + j;
        float s = s_scores[wave][sm_row][j];
        const bool ok = sm_valid && key_abs <= sm_abs_pos &&
                        key_abs < (long)kv_len;
        sc[jj] = ok ? s : -1e30f;
        tmax = fmaxf(tmax, sc[jj]);
      }
      tmax = fmaxf(tmax, __shfl_xor(tmax, 1, 64));
      tmax = fmaxf(tmax, __shfl_xor(tmax, 2, 64));
      float m_new = fmaxf(run_m, tmax);
      float alpha, rowsum = 0.f;
      if (m_new > -1e30f) {
        alpha = (run_m > -1e30f) ? __expf(run_m - m_new) : 0.f;
#pragma unroll
        for (int jj = 0; jj < 8; ++jj) {
          float p = (sc[jj] > -1e30f) ? __expf(sc[jj] - m_new) : 0.f;
          p_lds[wave][sm_row][sm_sub * 8 + jj] = f2bf(p);
          rowsum += p;
        }
      } else {
        alpha = 1.f;
#pragma unroll
        for (int jj = 0; jj < 8; ++jj)
          p_lds[wave][sm_row][sm_sub * 8 + jj] = 0;
      }
      rowsum += __shfl_xor(rowsum, 1, 64);
      rowsum += __shfl_xor(rowsum, 2, 64);
      run_l = run_l * alpha + rowsum;
      run_m = m_new;
      if (sm_sub == 0) {
        s_alpha[wave][sm_row] = alpha;
        s_rowl[wave][sm_row] = run_l;
      }
    }
    __builtin_amdgcn_s_waitcnt(0);
    __builtin_amdgcn_wave_barrier();

    // ---- rescale O and accumulate P*V
    {
      // rescale: C row = (lane>>4)*4 + i
      float al[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) al[i] = s_alpha[wave][(lane >> 4) * 4 + i];
      // P A-frag: row = lane&15, k(key) = (lane>>4)*8 + i
      bf16x8 pa0, pa1;
      {
        const int row = lane & 15;
        const int k0 = (lane >> 4) * 8;
#pragma unroll
        for (int i = 0; i < 8; ++i) pa0[i] = p_lds[wave][row][k0 + i];
        // second half keys 16..31 wait: KSTEP=32 = one mfma k=32 — single
        // A-frag covers keys (lane>>4)*8.. only 32 total: k index range is
        // 0..31 -> (lane>>4)*8+i covers 0..31. pa1 unused.
        pa1 = pa0;
      }
#pragma unroll
      for (int ct = 0; ct < CTILES; ++ct) {
#pragma unroll
        for (int i = 0; i < 4; ++i) o_acc[ct][i] *= al[i];
        // V B-frag: col = lane&15 -> dim = ct*16 + (lane&15),
        //           k(key) = (lane>>4)*8 + i -> strided column read
        bf16x8 vb;
        const int dim = ct * 16 + (lane & 15);
        const int kk0 = (lane >> 4) * 8;
#pragma unroll
        for (int i = 0; i < 8; ++i)
          vb[i] = v_lds[kk0 + i][swz_col(kk0 + i, dim)];
        o_acc[ct] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pa0, vb,
                                                            o_acc[ct], 0, 0, 0);
      }
    }
    __syncthreads();
  }

  // ---- epilogue: normalize + store
  {
    float invl[4];
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      float l = s_rowl[wave][(lane >> 4) * 4 + i];
      invl[i] = (l > 0.f) ? 1.f / l : 0.f;
    }
#pragma unroll
    for (int ct = 0; ct < CTILES; ++ct) {
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int row = (lane >> 4) * 4 + i;
        const int row_global = wave * 16 + row;
        const int pos_local = tile * POS_PER_WG + row_global / G;
        if (pos_local >= q_len) continue;
        const int head = row_global % G;
        const int tok = q_start + pos_local;
        const int dim = ct * 16 + (lane & 15);
        out[((long)tok * Hq + kvh * G + head) * D + dim] =
            f2bf(o_acc[ct][i] * invl[i]);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Host wrappers
// ---------------------------------------------------------------------------

void attn_decode_paged(torch::Tensor out, torch::Tensor q,
                       torch::Tensor kcache, torch::Tensor vcache,
                       torch::Tensor block_tables, torch::Tensor kv_lens,
                       double scale) {
  const int B = q.size(0), Hq = q.size(1), D = q.size(2);
  const int Hkv = kcache.size(1);
  const int G = Hq / Hkv;
  const int max_blocks = block_tables.size(1);
  TORCH_CHECK(kcache.size(2) == 16, "block_size must be 16");
  TORCH_CHECK(block_tables.scalar_type() == torch::kInt32);
  TORCH_CHECK(q.stride(2) == 1 && q.stride(1) == D, "q heads must be dense");
  TORCH_CHECK(out.is_contiguous(), "out must be contiguous");
  const long q_tstride = q.stride(0);
  // Fixed split count keeps the launch hipGraph-capturable while filling
  // the chip at small batch (B=1: 8 WGs -> 8*S*Hkv WGs) AND bounding the
  // serial page chain per wave (the decode critical path at long kv).
  // DTS_DECODE_SPLITS tunes it: the r2 bench kernel table showed the
  // combine kernel (whose work scales with S) at 6.7% of GPU busy while
  // typical bench kv is ~3k (6 pages/split at S=32 — launch-bound).
  // Measured on the flagship bench (r2): S=8 0.3549 traj/s, S=16
  // 0.3491, S=32 0.3266 — at the search's typical kv (~3k) 32 splits
  // left each workgroup ~6 pages of work (launch/latency-bound) and the
  // combine kernel at 6.7% of GPU busy. S=8 still gives 8*Hkv*B
  // workgroups (512 at B=8) and bounded the round-1 12k-kv probe fine.
  static const int S = [] {
    const char* e = getenv("DTS_DECODE_SPLITS");
    int v = e ? atoi(e) : 8;
    return (v >= 1 && v <= 64) ? v : 8;
  }();
  // Workspace cached per shape: layers within a step run sequentially on
  // one stream, so one buffer serves all 32 layer calls (and, being
  // allocated at warm-up time, lives OUTSIDE the graph pool).
  static std::unordered_map<std::string, std::array<torch::Tensor, 3>> ws_cache;
  const std::string key = std::to_string(q.get_device()) + ":" +
                          std::to_string(B) + ":" + std::to_string(Hkv) + ":" +
                          std::to_string(G) + ":" + std::to_string(D);
  auto it = ws_cache.find(key);
  if (it == ws_cache.end()) {
    auto f32 =
        torch::TensorOptions().dtype(torch::kFloat32).device(q.device());
    it = ws_cache
             .emplace(key,
                      std::array<torch::Tensor, 3>{
                          torch::empty({B, Hkv, S, G}, f32),
                          torch::empty({B, Hkv, S, G}, f32),
                          torch::empty({(long)B, Hkv, S, G, D}, f32)})
             .first;
  }
  torch::Tensor ws_m = it->second[0];
  torch::Tensor ws_l = it->second[1];
  torch::Tensor ws_o = it->second[2];
  dim3 grid(B, Hkv, S);
  dim3 block(256);
  auto stream = c10::hip::getCurrentHIPStream();
#define DECODE_CASE(g, d)                                                    \
  do {                                                                       \
    hipLaunchKernelGGL((attn_decode_kernel<g, d>), grid, block, 0, stream,   \
                       (float*)ws_m.data_ptr(), (float*)ws_l.data_ptr(),     \
                       (float*)ws_o.data_ptr(), (const short*)q.data_ptr(),  \
                       (const short*)kcache.data_ptr(),                      \
                       (const short*)vcache.data_ptr(),                      \
                       (const int*)block_tables.data_ptr(),                  \
                       (const int*)kv_lens.data_ptr(), max_blocks, Hkv,      \
                       (float)scale, q_tstride);                             \
    hipLaunchKernelGGL((attn_decode_combine<d>), dim3(B, Hq), dim3(256), 0,  \
                       stream, (short*)out.data_ptr(),                       \
                       (const float*)ws_m.data_ptr(),                        \
                       (const float*)ws_l.data_ptr(),                        \
                       (const float*)ws_o.data_ptr(), S, G, Hkv);            \
  } while (0)
  if (D == 128 && G == 4) DECODE_CASE(4, 128);
  else if (D == 128 && G == 8) DECODE_CASE(8, 128);
  else if (D == 128 && G == 1) DECODE_CASE(1, 128);
  else if (D == 64 && G == 1) DECODE_CASE(1, 64);
  else TORCH_CHECK(false, "unsupported decode attn shape D=", D, " G=", G);
#undef DECODE_CASE
  HIP_CHECK_LAST();
}

void attn_prefill_paged(torch::Tensor out, torch::Tensor q, torch::Tensor cu_q,
                        torch::Tensor q_pos, torch::Tensor kcache,
                        torch::Tensor vcache, torch::Tensor block_tables,
                        torch::Tensor kv_lens, double scale, int64_t swz) {
  // variant select via DTS_PREFILL_V: 0 -> v1 (KSTEP=32 LDS softmax),
  // 1 -> v1+XOR swizzle (-24..-30%), 2/3 -> v2 ablation (reg-staged +
  // transposed-V; SLOWER, profiles/prefill_v2_ablation_r2.md),
  // 4/5 -> v5 swapped-QK^T in-register softmax (KSTEP 32/64).
  // Default: v5 KSTEP=64 — measured 149/195/242 TF at N=2k/4k/8k vs
  // v1's 140/182/220 (PMC-guided: v1 parked 43% of wave cycles on the
  // softmax LDS round-trip fences).
  if (swz < 0) {
    static int env_v = [] {
      const char* e = getenv("DTS_PREFILL_V");
      return e ? atoi(e) : -1;
    }();
    swz = (env_v >= 0) ? env_v : 5;
  }
  const int Hq = q.size(1), D = q.size(2);
  const int Hkv = kcache.size(1);
  const int G = Hq / Hkv;
  const int P = cu_q.size(0) - 1;
  const int max_blocks = block_tables.size(1);
  TORCH_CHECK(q.stride(2) == 1 && q.stride(1) == D, "q heads must be dense");
  TORCH_CHECK(out.is_contiguous(), "out must be contiguous");
  const long q_tstride = q.stride(0);
  TORCH_CHECK(D == 128 || D == 64, "prefill kernel supports head_dim 64/128");
  TORCH_CHECK(kcache.size(2) == 16, "block_size must be 16");
  TORCH_CHECK(cu_q.scalar_type() == torch::kInt32);
  // worst-case tiles per seq: computed on host from q sizes is per-layer
  // overhead; use Tq (total) as bound and let tiles beyond q_len exit.
  const long Tq = q.size(0);
  const int pos_per_wg = 64 / G;
  const int max_tiles = (int)((Tq + pos_per_wg - 1) / pos_per_wg);
  dim3 grid(max_tiles, P, Hkv);
  dim3 block(256);
  auto stream = c10::hip::getCurrentHIPStream();
#define PREFILL_CASE(g, d)                                                    \
  do {                                                                        \
    if (swz == 2)                                                             \
      hipLaunchKernelGGL((attn_prefill_kernel_v2<g, d, 64>), grid, block, 0,  \
                         stream, (short*)out.data_ptr(),                      \
                         (const short*)q.data_ptr(),                          \
                         (const int*)cu_q.data_ptr(),                         \
                         (const long*)q_pos.data_ptr(),                       \
                         (const short*)kcache.data_ptr(),                     \
                         (const short*)vcache.data_ptr(),                     \
                         (const int*)block_tables.data_ptr(),                 \
                         (const int*)kv_lens.data_ptr(), max_blocks, Hkv,     \
                         (float)scale, q_tstride);                            \
    else if (swz == 3)                                                        \
      hipLaunchKernelGGL((attn_prefill_kernel_v2<g, d, 32>), grid, block, 0,  \
                         stream, (short*)out.data_ptr(),                      \
                         (const short*)q.data_ptr(),                          \
                         (const int*)cu_q.data_ptr(),                         \
                         (const long*)q_pos.data_ptr(),                       \
                         (const short*)kcache.data_ptr(),                     \
                         (const short*)vcache.data_ptr(),                     \
                         (const int*)block_tables.data_ptr(),                 \
                         (const int*)kv_lens.data_ptr(), max_blocks, Hkv,     \
                         (float)scale, q_tstride);                            \
    else if (swz == 4)                                                        \
      hipLaunchKernelGGL((attn_prefill_kernel_v5<g, d, 32>), grid, block, 0,  \
                         stream, (short*)out.data_ptr(),                      \
                         (const short*)q.data_ptr(),                         \
                         (const int*)cu_q.data_ptr(),                         \
                         (const long*)q_pos.data_ptr(),                       \
                         (const short*)kcache.data_ptr(),                     \
                         (const short*)vcache.data_ptr(),                     \
                         (const int*)block_tables.data_ptr(),                 \
                         (const int*)kv_lens.data_ptr(), max_blocks, Hkv,     \
                         (float)scale, q_tstride);                            \
    else if (swz == 5)                                                        \
      hipLaunchKernelGGL((attn_prefill_kernel_v5<g, d, 64>), grid, block, 0,  \
                         stream, (short*)out.data_ptr(),                      \
                         (const short*)q.data_ptr(),                         \
                         (const int*)cu_q.data_ptr(),                         \
                         (const long*)q_pos.data_ptr(),                       \
                         (const short*)kcache.data_ptr(),                     \
                         (const short*)vcache.data_ptr(),                     \
                         (const int*)block_tables.data_ptr(),                 \
                         (const int*)kv_lens.data_ptr(), max_blocks, Hkv,     \
                         (float)scale, q_tstride);                            \
    else if (swz == 1)                                                        \
      hipLaunchKernelGGL((attn_prefill_kernel<g, d, true>), grid, block, 0,   \
                         stream, (short*)out.data_ptr(),                      \
                         (const short*)q.data_ptr(),                          \
                         (const int*)cu_q.data_ptr(),                         \
                         (const long*)q_pos.data_ptr(),                       \
                         (const short*)kcache.data_ptr(),                     \
                         (const short*)vcache.data_ptr(),                     \
                         (const int*)block_tables.data_ptr(),                 \
                         (const int*)kv_lens.data_ptr(), max_blocks, Hkv,     \
                         (float)scale, q_tstride);                            \
    else                                                                      \
      hipLaunchKernelGGL((attn_prefill_kernel<g, d, false>), grid, block, 0,  \
                         stream, (short*)out.data_ptr(),                      \
                         (const short*)q.data_ptr(),                          \
                         (const int*)cu_q.data_ptr(),                         \
                         (const long*)q_pos.data_ptr(),                       \
                         (const short*)kcache.data_ptr(),                     \
                         (const short*)vcache.data_ptr(),                     \
                         (const int*)block_tables.data_ptr(),                 \
                         (const int*)kv_lens.data_ptr(), max_blocks, Hkv,     \
                         (float)scale, q_tstride);                            \
  } while (0)
  if (D == 128 && G == 4) PREFILL_CASE(4, 128);
  else if (D == 128 && G == 8) PREFILL_CASE(8, 128);
  else if (D == 128 && G == 1) PREFILL_CASE(1, 128);
  else if (D == 128 && G == 2) PREFILL_CASE(2, 128);
  else if (D == 64 && G == 1) PREFILL_CASE(1, 64);
  else TORCH_CHECK(false, "unsupported prefill shape G=", G, " D=", D);
#undef PREFILL_CASE
  HIP_CHECK_LAST();
}
