// Varlen causal prefill attention for gfx950 — flash-style MFMA kernel.
//
// Structure (cdna_hip_programming.md Appendix B "Fused attention prefill"):
//   - an 8-WAVE workgroup serves 8 consecutive 32-row q-tiles of ONE
//     (sequence, q-head): the K/V LDS staging is shared by the 8 waves, so
//     its cost is amortized 8x over the 1-wave-per-block version;
//   - KV tiles of 32 keys staged in LDS; K XOR-swizzled (G4: row-major tiles
//     at D=128 are an up-to-16-way ds_read_b128 conflict; byte ^= (row&15)<<4
//     makes the 16-lane group conflict-free); V staged TRANSPOSED
//     ([128][32+pad]) so the PV A-fragment reads are contiguous 16B;
//   - SWAPPED QK^T: S = mfma(A=K_tile, B=Q) so each lane's accumulators hold
//     one q-row's scores -> softmax is in-register (fmax chain + one
//     permlane32_swap half-merge), no cross-lane LDS traffic;
//   - P repacked to bf16 MFMA B-fragments with pack + permlane32_swap half
//     exchanges (T12/T21 pattern);
//   - PV as O^T = mfma(A=V^T, B=P) over 4 d-tiles;
//   - online softmax with per-lane running max/sum (rows are lane-local).
//
// MFMA fragment maps (v_mfma_f32_32x32x16_bf16, verified on-device by the
// mfma_selftest binding):
//   A[i][k]: lane l holds i = l&31, k = (l>>5)*8 + j   (j = 0..7)
//   B[k][j]: lane l holds j = l&31, k = (l>>5)*8 + jj
//   C/D[i][j]: lane l holds j = l&31, i = (r&3) + 8*(r>>2) + 4*(l>>5)  (r = 0..15)
//
// Host metadata (built in ops/__init__._prefill_tiles): q-tiles grouped 4 per
// block PER SEQUENCE; grp_qpos0 = -1 pads idle waves.

#include "common.h"

typedef __bf16 bf16x8_mfma __attribute__((ext_vector_type(8)));

#define QBLK 32
#define KVBLK 64                   // staged keys per barrier round (2 MFMA sub-tiles)
#define DHEAD 128
#define NWAVES 8
#define K_ROW_BYTES 256            // 128 bf16
#define VT_ROW_SHORTS 72           // 64 keys + 8 pad (conflict-free b128 reads)

__device__ __forceinline__ int acc_row(int r, int half) {
  return (r & 3) + 8 * (r >> 2) + 4 * half;   // C/D row map
}

extern "C" __global__ void __launch_bounds__(64 * NWAVES) attn_prefill_kernel(
    bf16_t* __restrict__ out,       // [T, H, D] contiguous
    const bf16_t* __restrict__ q,   // [T, H, D], token stride q_tstride
    const bf16_t* __restrict__ k,   // [T, KVH, D], token stride kv_tstride
    const bf16_t* __restrict__ v,
    const int* __restrict__ grp_seq_start,   // [G] first token of the group's seq
    const int* __restrict__ grp_seqlen,      // [G]
    const int* __restrict__ grp_qpos0,       // [G*NWAVES], -1 = idle wave
    float scale, int num_q_heads, int num_kv_heads, int q_tstride, int kv_tstride) {
  const int grp = blockIdx.x;
  const int h = blockIdx.y;
  const int g_kv = h / (num_q_heads / num_kv_heads);
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int half = lane >> 5;
  const int col = lane & 31;       // q-row (B/C col) or K/VT row (A row)

  const int seq_start = grp_seq_start[grp];
  const int seqlen = grp_seqlen[grp];
  const int qpos0 = grp_qpos0[grp * NWAVES + wave];   // -1: idle wave
  const bool active = qpos0 >= 0;
  const int my_qpos = active ? qpos0 + col : 0;

  __shared__ __attribute__((aligned(16))) char smem[KVBLK * K_ROW_BYTES +
                                                    DHEAD * VT_ROW_SHORTS * 2];
  char* k_lds = smem;                                   // [64][256B], XOR-swizzled
  short* vt_lds = reinterpret_cast<short*>(smem + KVBLK * K_ROW_BYTES);  // [128][72]

  // ---- Q fragments (B-operand): 8 d-slices of 16 --------------------------
  bf16x8_mfma q_frag[8];
  {
    const int qrow_c = active ? min(my_qpos, seqlen - 1) : 0;
    const bf16_t* qp = q + (int64_t)(seq_start + qrow_c) * q_tstride + h * DHEAD;
#pragma unroll
    for (int s = 0; s < 8; ++s)
      q_frag[s] = *reinterpret_cast<const bf16x8_mfma*>((const short*)qp + s * 16 + half * 8);
  }

  f32x16 o_acc[4] = {};
  float m_run = -INFINITY;
  float l_run = 0.0f;

  // group-wide kv bound: the group's last active wave reaches furthest
  int kv_end_grp = 0;
#pragma unroll
  for (int w = 0; w < NWAVES; ++w) {
    const int qp0 = grp_qpos0[grp * NWAVES + w];
    if (qp0 >= 0) kv_end_grp = max(kv_end_grp, min(seqlen, qp0 + QBLK));
  }
  const int kv_end_me = active ? min(seqlen, qpos0 + QBLK) : 0;

  for (int kt00 = 0; kt00 < kv_end_grp; kt00 += KVBLK) {
    const int nkeys_blk = min(KVBLK, kv_end_grp - kt00);

    // ---- stage K (swizzled) + V^T, all 512 threads cooperatively ----------
    {
      const int key = tid >> 3;               // one key per 8 threads
      const int piece = tid & 7;              // 16-dim sliver of that key
      const bool valid = key < nkeys_blk;
      const bf16_t* kp = k + (int64_t)(seq_start + kt00 + min(key, nkeys_blk - 1)) * kv_tstride + g_kv * DHEAD;
#pragma unroll
      for (int c = 0; c < 2; ++c) {
        const int byte_off = piece * 32 + c * 16;
        bf16x8_vec val = valid
            ? reinterpret_cast<const bf16x8_vec*>((const short*)kp)[byte_off / 16]
            : bf16x8_vec{0, 0, 0, 0, 0, 0, 0, 0};
        *reinterpret_cast<bf16x8_vec*>(
            k_lds + key * K_ROW_BYTES + (byte_off ^ ((key & 15) << 4))) = val;
      }
      // V^T: this thread transposes its [16 dims x 1 key] sliver
      const bf16_t* vp = v + (int64_t)(seq_start + kt00 + min(key, nkeys_blk - 1)) * kv_tstride + g_kv * DHEAD;
      bf16x8_vec va = valid ? reinterpret_cast<const bf16x8_vec*>((const short*)vp)[piece * 2]
                            : bf16x8_vec{0, 0, 0, 0, 0, 0, 0, 0};
      bf16x8_vec vb = valid ? reinterpret_cast<const bf16x8_vec*>((const short*)vp)[piece * 2 + 1]
                            : bf16x8_vec{0, 0, 0, 0, 0, 0, 0, 0};
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        vt_lds[(piece * 16 + j) * VT_ROW_SHORTS + key] = va[j];
        vt_lds[(piece * 16 + 8 + j) * VT_ROW_SHORTS + key] = vb[j];
      }
    }
    __syncthreads();

#pragma unroll
    for (int sub = 0; sub < 2; ++sub) {
    const int kt0 = kt00 + sub * 32;
    const int nkeys = min(32, nkeys_blk - sub * 32);
    if (active && kt0 < kv_end_me && nkeys > 0) {
      // ---- QK^T: S[key][qrow] over 8 d-slices -----------------------------
      f32x16 s_acc = {};
#pragma unroll
      for (int s = 0; s < 8; ++s) {
        const int byte_off = (s * 2 + half) * 16;
        const int krow = col + sub * 32;
        bf16x8_mfma a_frag = *reinterpret_cast<const bf16x8_mfma*>(
            k_lds + krow * K_ROW_BYTES + (byte_off ^ ((krow & 15) << 4)));
        s_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a_frag, q_frag[s], s_acc, 0, 0, 0);
      }

      // ---- mask + scale ----------------------------------------------------
      float sv[16];
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int key = kt0 + acc_row(r, half);
        const bool ok = (key <= my_qpos) && (acc_row(r, half) < nkeys) && (key < seqlen);
        sv[r] = ok ? s_acc[r] * scale : -INFINITY;
      }

      // ---- online softmax (row = lane-local) -------------------------------
      float tmax = sv[0];
#pragma unroll
      for (int r = 1; r < 16; ++r) tmax = fmaxf(tmax, sv[r]);
      {
        auto sw = __builtin_amdgcn_permlane32_swap(__float_as_uint(tmax), __float_as_uint(tmax), false, false);
        tmax = fmaxf(__uint_as_float(sw[0]), __uint_as_float(sw[1]));
      }
      // defer-max (T13): rows are lane-local, so the defer decision is
      // per-lane — keep the old running max while the tile max grew < THR
      // (P then bounded by e^THR, fine for f32 p / bf16 pack). Decision
      // happens BEFORE this tile's P is exponentiated (the safe order).
      const float DEFER_THR = 8.0f;
      float m_new, alpha;
      if (tmax <= m_run + DEFER_THR) {       // first tile: m_run=-inf fails this
        m_new = m_run;
        alpha = 1.0f;
      } else {
        m_new = fmaxf(m_run, tmax);
        alpha = (m_run == -INFINITY) ? 0.0f : __expf(m_run - m_new);
      }
      m_run = m_new;
      if (__builtin_amdgcn_ballot_w64(alpha != 1.0f)) {
#pragma unroll
        for (int dt = 0; dt < 4; ++dt)
#pragma unroll
          for (int r = 0; r < 16; ++r) o_acc[dt][r] *= alpha;
      }

      float p[16];
      float psum = 0.0f;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        p[r] = (sv[r] == -INFINITY) ? 0.0f : __expf(sv[r] - m_new);
        psum += p[r];
      }
      l_run = l_run * alpha + psum;

      // ---- repack P -> bf16 B-fragments (T12/T21 half swaps) ---------------
      uint32_t packs[8];
#pragma unroll
      for (int pr = 0; pr < 8; ++pr) {
        const short lo = f32_to_bf16(p[2 * pr]);
        const short hi = f32_to_bf16(p[2 * pr + 1]);
        packs[pr] = ((uint32_t)(uint16_t)hi << 16) | (uint16_t)lo;
      }
      uint32_t bfrag[2][4];
#pragma unroll
      for (int kt = 0; kt < 2; ++kt) {
        auto r0 = __builtin_amdgcn_permlane32_swap(packs[4 * kt + 0], packs[4 * kt + 2], false, false);
        auto r1 = __builtin_amdgcn_permlane32_swap(packs[4 * kt + 1], packs[4 * kt + 3], false, false);
        bfrag[kt][0] = r0[0];
        bfrag[kt][1] = r1[0];
        bfrag[kt][2] = r0[1];
        bfrag[kt][3] = r1[1];
      }

      // ---- PV: O^T += V^T x P ----------------------------------------------
#pragma unroll
      for (int dt = 0; dt < 4; ++dt) {
        const int d_row = dt * 32 + col;
#pragma unroll
        for (int kt = 0; kt < 2; ++kt) {
          bf16x8_mfma vfrag = *reinterpret_cast<const bf16x8_mfma*>(
              vt_lds + d_row * VT_ROW_SHORTS + sub * 32 + kt * 16 + half * 8);
          bf16x8_mfma pfrag;
          {
            uint32_t* pf = reinterpret_cast<uint32_t*>(&pfrag);
#pragma unroll
            for (int w = 0; w < 4; ++w) pf[w] = bfrag[kt][w];
          }
          o_acc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vfrag, pfrag, o_acc[dt], 0, 0, 0);
        }
      }
    }
    }  // sub
    __syncthreads();
  }

  // ---- epilogue: merge half-sums, normalize, write O ----------------------
  {
    auto sw = __builtin_amdgcn_permlane32_swap(__float_as_uint(l_run), __float_as_uint(l_run), false, false);
    l_run = __uint_as_float(sw[0]) + __uint_as_float(sw[1]);
  }
  if (active && my_qpos < seqlen) {
    const float inv_l = (l_run > 0.0f) ? 1.0f / l_run : 0.0f;
    bf16_t* op = out + (((int64_t)(seq_start + my_qpos)) * num_q_heads + h) * DHEAD;
#pragma unroll
    for (int dt = 0; dt < 4; ++dt) {
#pragma unroll
      for (int g2 = 0; g2 < 4; ++g2) {   // regs 4*g2..+3 = d consecutive
        const int d0 = dt * 32 + 8 * g2 + 4 * half;
        uint64_t word = 0;
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          const float val = o_acc[dt][4 * g2 + j] * inv_l;
          word |= ((uint64_t)(uint16_t)f32_to_bf16(val)) << (16 * j);
        }
        *reinterpret_cast<uint64_t*>((short*)op + d0) = word;
      }
    }
  }
}
