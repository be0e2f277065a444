// Varlen causal prefill attention for gfx950 — flash-style MFMA kernel.
//
// Structure (cdna_hip_programming.md Appendix B "Fused attention prefill",
// simplified to a 1-wave workgroup):
//   - one wave per 32 q-rows of one (sequence, q-head); grid covers all
//     q-tiles x heads;
//   - KV tiles of 32 keys staged in LDS; K XOR-swizzled (G4: row-major tiles
//     at D=128 are an up-to-16-way ds_read_b128 conflict; byte ^= (row&15)<<4
//     makes the 16-lane group conflict-free);
//   - SWAPPED QK^T: S = mfma(A=K_tile, B=Q) so each lane's accumulators hold
//     one q-row's scores -> softmax is in-register (fmax chain + one
//     permlane32_swap half-merge), no cross-lane LDS traffic;
//   - P repacked to bf16 MFMA B-fragments with v_cvt_pk-style packing + half
//     swaps (T12/T21 pattern);
//   - PV as O^T = mfma(A=V^T, B=P) over 4 d-tiles; V^T staged in LDS (b16
//     transpose writes — the v1 cost; tr_b16 reads are the planned upgrade);
//   - online softmax with per-lane running max/sum (rows are lane-local).
//
// MFMA fragment maps assumed (v_mfma_f32_32x32x16_bf16, verified on-device by
// the mfma_selftest binding):
//   A[i][k]: lane l holds i = l&31, k = (l>>5)*8 + j   (j = 0..7)
//   B[k][j]: lane l holds j = l&31, k = (l>>5)*8 + jj
//   C/D[i][j]: lane l holds j = l&31, i = (r&3) + 8*(r>>2) + 4*(l>>5)  (r = 0..15)

#include "common.h"

typedef __bf16 bf16x8_mfma __attribute__((ext_vector_type(8)));

#define QBLK 32
#define KVBLK 32
#define DHEAD 128
#define K_ROW_BYTES 256           // 128 bf16
#define VT_ROW_BYTES 80           // 32 keys * 2B + 16B pad (conflict-free reads)

__device__ __forceinline__ int acc_row(int r, int half) {
  return (r & 3) + 8 * (r >> 2) + 4 * half;   // C/D row map
}

extern "C" __global__ void __launch_bounds__(64) attn_prefill_kernel(
    bf16_t* __restrict__ out,       // [T, H, D]
    const bf16_t* __restrict__ q,   // [T, H, D]
    const bf16_t* __restrict__ k,   // [T, KVH, D]
    const bf16_t* __restrict__ v,   // [T, KVH, D]
    const int* __restrict__ tile_seq_start,  // per q-tile: first token of seq
    const int* __restrict__ tile_qpos0,     // per q-tile: row0 position in seq
    const int* __restrict__ tile_seqlen,    // per q-tile: sequence length
    float scale, int num_q_heads, int num_kv_heads, int q_tstride, int kv_tstride) {
  const int tile = blockIdx.x;
  const int h = blockIdx.y;
  const int g_kv = h / (num_q_heads / num_kv_heads);
  const int lane = threadIdx.x;
  const int half = lane >> 5;      // 0 | 1
  const int col = lane & 31;       // q-row (B/C col) or K/VT row (A row)

  const int seq_start = tile_seq_start[tile];
  const int qpos0 = tile_qpos0[tile];
  const int seqlen = tile_seqlen[tile];
  const int my_qpos = qpos0 + col;

  __shared__ __attribute__((aligned(16))) char smem[KVBLK * K_ROW_BYTES + DHEAD * VT_ROW_BYTES];
  char* k_lds = smem;                              // [32][256B], XOR-swizzled
  char* vt_lds = smem + KVBLK * K_ROW_BYTES;       // [128][80B] = V^T

  // ---- load Q fragments (B-operand): 8 d-slices of 16 ---------------------
  // lane l: Q[qpos0 + col][s*16 + half*8 + 0..7], pre-scaled into bf16? No:
  // keep bf16 raw; fold `scale` into the softmax exp argument instead.
  bf16x8_mfma q_frag[8];
  {
    const int qrow_clamped = min(my_qpos, seqlen - 1);   // tail rows clamped (never written out)
    const bf16_t* qp = q + (int64_t)(seq_start + qrow_clamped) * q_tstride + h * DHEAD;
#pragma unroll
    for (int s = 0; s < 8; ++s)
      q_frag[s] = *reinterpret_cast<const bf16x8_mfma*>((const short*)qp + s * 16 + half * 8);
  }

  f32x16 o_acc[4] = {};            // O^T accumulators: d-tiles of 32
  float m_run = -INFINITY;         // running max of SCALED scores (this lane's q-row)
  float l_run = 0.0f;              // running half-sum of P

  const int kv_end = min(seqlen, qpos0 + QBLK);   // causal upper bound
  for (int kt0 = 0; kt0 < kv_end; kt0 += KVBLK) {
    const int nkeys = min(KVBLK, kv_end - kt0);

    // ---- stage K tile (swizzled rows) and V^T tile ------------------------
    // K: lane covers key=col, 128B half: bytes [half*128, half*128+128)
    {
      const int key = col;
      const bool valid = key < nkeys;
      const bf16_t* kp = k + (int64_t)(seq_start + kt0 + min(key, nkeys - 1)) * kv_tstride + g_kv * DHEAD;
#pragma unroll
      for (int c = 0; c < 8; ++c) {
        const int byte_off = half * 128 + c * 16;
        bf16x8_vec val = valid
            ? reinterpret_cast<const bf16x8_vec*>((const short*)kp)[byte_off / 16]
            : bf16x8_vec{0, 0, 0, 0, 0, 0, 0, 0};
        *reinterpret_cast<bf16x8_vec*>(
            k_lds + key * K_ROW_BYTES + (byte_off ^ ((key & 15) << 4))) = val;
      }
      // V^T: lane loads V[key][half*64 .. half*64+63] and scatters b16 writes
      const bf16_t* vp = v + (int64_t)(seq_start + kt0 + min(key, nkeys - 1)) * kv_tstride + g_kv * DHEAD;
#pragma unroll
      for (int c = 0; c < 8; ++c) {
        bf16x8_vec val = valid
            ? reinterpret_cast<const bf16x8_vec*>((const short*)vp)[half * 8 + c]
            : bf16x8_vec{0, 0, 0, 0, 0, 0, 0, 0};
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const int d = half * 64 + c * 8 + j;
          *reinterpret_cast<short*>(vt_lds + d * VT_ROW_BYTES + key * 2) = val[j];
        }
      }
    }
    __syncthreads();   // single wave: compiles to the needed lgkmcnt wait

    // ---- QK^T: S[key][qrow] over 8 d-slices -------------------------------
    f32x16 s_acc = {};
#pragma unroll
    for (int s = 0; s < 8; ++s) {
      const int key_row = col;     // A row
      const int byte_off = (s * 2 + half) * 16;
      bf16x8_mfma a_frag = *reinterpret_cast<const bf16x8_mfma*>(
          k_lds + key_row * K_ROW_BYTES + (byte_off ^ ((key_row & 15) << 4)));
      s_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a_frag, q_frag[s], s_acc, 0, 0, 0);
    }

    // ---- mask + scale ------------------------------------------------------
    float sv[16];
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int key = kt0 + acc_row(r, half);
      const bool ok = (key < kv_end) && (key <= my_qpos) && (acc_row(r, half) < nkeys);
      sv[r] = ok ? s_acc[r] * scale : -INFINITY;
    }

    // ---- online softmax (row = lane-local) --------------------------------
    float tmax = sv[0];
#pragma unroll
    for (int r = 1; r < 16; ++r) tmax = fmaxf(tmax, sv[r]);
    {
      auto sw = __builtin_amdgcn_permlane32_swap(__float_as_uint(tmax), __float_as_uint(tmax), false, false);
      tmax = fmaxf(__uint_as_float(sw[0]), __uint_as_float(sw[1]));
    }
    const float m_new = fmaxf(m_run, tmax);
    const float alpha = (m_run == -INFINITY) ? 0.0f : __expf(m_run - m_new);
    m_run = m_new;
#pragma unroll
    for (int dt = 0; dt < 4; ++dt)
#pragma unroll
      for (int r = 0; r < 16; ++r) o_acc[dt][r] *= alpha;

    float p[16];
    float psum = 0.0f;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      p[r] = (sv[r] == -INFINITY) ? 0.0f : __expf(sv[r] - m_new);
      psum += p[r];
    }
    l_run = l_run * alpha + psum;

    // ---- repack P -> bf16 B-fragments (key-major) -------------------------
    // pack pairs (r, r+1) -> keys (4h + r&3 pair); swap halves to build
    // B[k = half*8 + j][qrow]: see T12/T21.
    uint32_t packs[8];
#pragma unroll
    for (int pr = 0; pr < 8; ++pr) {
      const short lo = f32_to_bf16(p[2 * pr]);
      const short hi = f32_to_bf16(p[2 * pr + 1]);
      packs[pr] = ((uint32_t)(uint16_t)hi << 16) | (uint16_t)lo;
    }
    uint32_t bfrag[2][4];  // [key-tile kt(0: keys 0-15, 1: keys 16-31)][4 u32]
#pragma unroll
    for (int kt = 0; kt < 2; ++kt) {
      auto r0 = __builtin_amdgcn_permlane32_swap(packs[4 * kt + 0], packs[4 * kt + 2], false, false);
      auto r1 = __builtin_amdgcn_permlane32_swap(packs[4 * kt + 1], packs[4 * kt + 3], false, false);
      bfrag[kt][0] = r0[0];
      bfrag[kt][1] = r1[0];
      bfrag[kt][2] = r0[1];
      bfrag[kt][3] = r1[1];
    }

    // ---- PV: O^T += V^T x P ------------------------------------------------
#pragma unroll
    for (int dt = 0; dt < 4; ++dt) {
      const int d_row = dt * 32 + col;
#pragma unroll
      for (int kt = 0; kt < 2; ++kt) {
        bf16x8_mfma vfrag = *reinterpret_cast<const bf16x8_mfma*>(
            vt_lds + d_row * VT_ROW_BYTES + (kt * 16 + half * 8) * 2);
        bf16x8_mfma pfrag;
        {
          uint32_t* pf = reinterpret_cast<uint32_t*>(&pfrag);
#pragma unroll
          for (int w = 0; w < 4; ++w) pf[w] = bfrag[kt][w];
        }
        o_acc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vfrag, pfrag, o_acc[dt], 0, 0, 0);
      }
    }
    __syncthreads();
  }

  // ---- epilogue: merge half-sums, normalize, write O ----------------------
  {
    auto sw = __builtin_amdgcn_permlane32_swap(__float_as_uint(l_run), __float_as_uint(l_run), false, false);
    l_run = __uint_as_float(sw[0]) + __uint_as_float(sw[1]);
  }
  if (my_qpos < seqlen && col < QBLK) {
    const float inv_l = (l_run > 0.0f) ? 1.0f / l_run : 0.0f;
    bf16_t* op = out + (((int64_t)(seq_start + my_qpos)) * num_q_heads + h) * DHEAD;
#pragma unroll
    for (int dt = 0; dt < 4; ++dt) {
#pragma unroll
      for (int grp = 0; grp < 4; ++grp) {   // regs 4*grp .. 4*grp+3 = d consecutive
        const int d0 = dt * 32 + 8 * grp + 4 * half;
        uint64_t word = 0;
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          const float val = o_acc[dt][4 * grp + j] * inv_l;
          word |= ((uint64_t)(uint16_t)f32_to_bf16(val)) << (16 * j);
        }
        *reinterpret_cast<uint64_t*>((short*)op + d0) = word;
      }
    }
  }
}
