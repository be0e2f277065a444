// Paged decode attention (single new token per stream) for gfx950.
//
// Memory-bound: the cost is streaming each sequence's K/V pages once
// (Appendix B "Attention decode"). One 256-thread block per
// (sequence, kv_head); the block serves ALL q-heads of that GQA group so K/V
// are read exactly once per sequence. Tiles of 256 keys; per tile:
//   phase A: score s[g][key] = q[g] . k[key]  (8-lane thread-groups per key,
//            16-byte K loads, shuffle-reduced)
//   phase B: block-wide online-softmax update (running max / sum, output rescale)
//   phase C: o[g][d] += p[key] * v[key][d]    (lane owns (g, d-pair), 4-byte
//            V loads; a V row is one coalesced 256 B read per head-group)
//
// Layouts: q [B, H, D]; caches [NB, KVH, BS, D]; block_tables [B, MAXB];
// context_lens INCLUDE the current token (its K/V are already stored).
// D = 128; GQA group size <= 8.

#include "common.h"

#define TKV 256            // keys per tile
#define NTHREADS 256
#define KLANES 8           // lanes cooperating on one key's dot product

extern "C" __global__ void __launch_bounds__(NTHREADS) attn_decode_kernel(
    bf16_t* __restrict__ out,            // [B, H, D]
    const bf16_t* __restrict__ q,        // [B, H, D]
    const bf16_t* __restrict__ k_cache,  // [NB, KVH, BS, D]
    const bf16_t* __restrict__ v_cache,
    const int* __restrict__ block_tables,  // [B, max_blocks]
    const int* __restrict__ context_lens,  // [B]
    float scale, int num_q_heads, int num_kv_heads, int head_dim,
    int block_size, int max_blocks) {
  const int b = blockIdx.x;
  const int g_kv = blockIdx.y;             // kv head
  const int gqa = num_q_heads / num_kv_heads;
  const int L = context_lens[b];
  const int tid = threadIdx.x;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* q_lds = reinterpret_cast<float*>(smem);          // [gqa][128]
  float* s_lds = q_lds + gqa * 128;                       // [gqa][TKV]
  float* red = s_lds + gqa * TKV;                         // [16] scratch
  float* mstate = red + 16;                               // [gqa] running max
  float* lstate = mstate + 8;                             // [gqa] running sum
  float* alpha_lds = lstate + 8;                          // [gqa] rescale

  // load q (scaled) into LDS as fp32
  for (int i = tid; i < gqa * 128; i += NTHREADS) {
    int g = i / 128, d = i % 128;
    const bf16_t* qp = q + (((int64_t)b * num_q_heads) + g_kv * gqa + g) * head_dim;
    q_lds[i] = bf16_to_f32(((const short*)qp)[d]) * scale;
  }
  if (tid < 8) { mstate[tid] = -INFINITY; lstate[tid] = 0.0f; }
  __syncthreads();

  // per-thread output accumulators: lane owns (g = tid/64, d pair within head)
  // each thread owns gqa_per_slot = ceil(gqa*128/ (NTHREADS*2)) handled below:
  // fixed mapping: thread t owns head-group g_o = t >> 6 (0..3), dims
  // d0 = (t & 63) * 2. For gqa > 4, each thread additionally owns g_o + 4.
  float o_acc[2][2];                      // [g-slot][2 dims]
#pragma unroll
  for (int a = 0; a < 2; ++a) { o_acc[a][0] = 0.f; o_acc[a][1] = 0.f; }

  const int* bt = block_tables + (int64_t)b * max_blocks;

  for (int tile = 0; tile < L; tile += TKV) {
    const int nkeys = min(TKV, L - tile);
    // ---- phase A: scores -------------------------------------------------
    // thread-group of KLANES lanes per key: NTHREADS/KLANES = 32 keys per pass
    const int kg = tid / KLANES;       // key index within pass group
    const int kl = tid % KLANES;       // lane within key group
    for (int key0 = 0; key0 < nkeys; key0 += NTHREADS / KLANES) {
      const int key = key0 + kg;
      if (key < nkeys) {
        const int gk = tile + key;
        const int64_t blk = bt[gk / block_size];
        const bf16_t* kp = k_cache +
            (((blk * num_kv_heads) + g_kv) * block_size + (gk % block_size)) * head_dim;
        // each lane loads 16 bf16 (32B): dims [kl*16, kl*16+16)
        const bf16x8_vec* kv8 = reinterpret_cast<const bf16x8_vec*>(kp) + kl * 2;
        bf16x8_vec ka = kv8[0];
        bf16x8_vec kb = kv8[1];
        for (int g = 0; g < gqa; ++g) {
          const float* qg = q_lds + g * 128 + kl * 16;
          float acc = 0.f;
#pragma unroll
          for (int j = 0; j < 8; ++j) acc += bf16_to_f32(ka[j]) * qg[j];
#pragma unroll
          for (int j = 0; j < 8; ++j) acc += bf16_to_f32(kb[j]) * qg[8 + j];
          // reduce across the 8-lane group (lanes kl = 0..7 contiguous)
#pragma unroll
          for (int off = 4; off > 0; off >>= 1) acc += __shfl_xor(acc, off);
          if (kl == 0) s_lds[g * TKV + key] = acc;
        }
      }
    }
    __syncthreads();

    // ---- phase B: online softmax update ---------------------------------
    for (int g = 0; g < gqa; ++g) {
      float lm = -INFINITY;
      for (int i = tid; i < nkeys; i += NTHREADS) lm = fmaxf(lm, s_lds[g * TKV + i]);
      float tile_max = block_reduce_max<4>(lm, red);
      float m_old, m_new;
      if (tid == 0) {
        m_old = mstate[g];
        m_new = fmaxf(m_old, tile_max);
        alpha_lds[g] = __expf(m_old - m_new);
        mstate[g] = m_new;
      }
      __syncthreads();
      m_new = mstate[g];
      float ls = 0.f;
      for (int i = tid; i < nkeys; i += NTHREADS) {
        float p = __expf(s_lds[g * TKV + i] - m_new);
        s_lds[g * TKV + i] = p;
        ls += p;
      }
      float tile_sum = block_reduce_sum<4>(ls, red);
      if (tid == 0) lstate[g] = lstate[g] * alpha_lds[g] + tile_sum;
      __syncthreads();
    }

    // ---- phase C: V accumulation ----------------------------------------
    const int d0 = (tid & 63) * 2;
    for (int slot = 0; slot < 2; ++slot) {
      const int g = (tid >> 6) + slot * 4;
      if (g >= gqa) break;
      const float alpha = alpha_lds[g];
      o_acc[slot][0] *= alpha;
      o_acc[slot][1] *= alpha;
      for (int key = 0; key < nkeys; ++key) {
        const float p = s_lds[g * TKV + key];
        const int gk = tile + key;
        const int64_t blk = bt[gk / block_size];
        const bf16_t* vp = v_cache +
            (((blk * num_kv_heads) + g_kv) * block_size + (gk % block_size)) * head_dim;
        const uint32_t pair = *reinterpret_cast<const uint32_t*>((const short*)vp + d0);
        o_acc[slot][0] += p * bf16_to_f32((short)(pair & 0xffff));
        o_acc[slot][1] += p * bf16_to_f32((short)(pair >> 16));
      }
    }
    __syncthreads();
  }

  // ---- epilogue: normalize and write -------------------------------------
  const int d0 = (tid & 63) * 2;
  for (int slot = 0; slot < 2; ++slot) {
    const int g = (tid >> 6) + slot * 4;
    if (g >= gqa) break;
    const float inv_l = 1.0f / lstate[g];
    bf16_t* op = out + (((int64_t)b * num_q_heads) + g_kv * gqa + g) * head_dim;
    uint32_t pair = ((uint32_t)(uint16_t)f32_to_bf16(o_acc[slot][1] * inv_l) << 16) |
                    (uint16_t)f32_to_bf16(o_acc[slot][0] * inv_l);
    *reinterpret_cast<uint32_t*>((short*)op + d0) = pair;
  }
}
