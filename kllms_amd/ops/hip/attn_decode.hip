// Paged decode attention (single new token per stream) for gfx950.
//
// Flash-decoding structure: the context is partitioned into key chunks and
// each (sequence, kv_head, chunk) is one 256-thread workgroup — with B x KVH
// blocks alone (e.g. 20 x 8 = 160) the 256-CU chip sits idle, so the chunk
// split is what fills it. Each block emits a partial (m, l, o[gqa][128]);
// a small reduce kernel merges chunks with the standard log-sum-exp combine.
//
// Memory behavior: each block streams its chunk of K/V exactly ONCE:
//   - scores: 8-lane thread groups per key, 16-byte K loads, in-wave reduce;
//   - online softmax: one wave per head (in-wave shuffles, single barrier);
//   - V accumulation: threads partition the KEYS (4 partitions x 64 dim-pair
//     lanes), each thread accumulating ALL q-heads for its dim pair, so a V
//     row is loaded once per block (a head-partitioned layout re-fetches the
//     64 KB V tile per head-group wave — it exceeds the 32 KB L1);
//   - per-key cache-row offsets staged in LDS once per tile.
//
// The kernel is TEMPLATED on the GQA group size: the per-head accumulator
// array must be statically indexed (a runtime-gqa loop sends it to scratch —
// cdna guide §5.4 rule 20) and the head loops fully unrolled.
//
// Layouts: q [B, H, D=128]; caches [NB, KVH, BS, D]; block_tables [B, MAXB];
// context_lens INCLUDE the current token. GQA in {1, 2, 4, 8}.

#include "common.h"

#define TKV ATTN_DECODE_TKV
#define NTHREADS 256
#define KLANES 4           // lanes cooperating on one key's dot product
// chunk size is a runtime arg (adaptive split-K: big batches need no split)

// partials layout: [B, KVH, max_chunks, gqa, 130]: 128 o values + m + l
#define PART_STRIDE 130

template <int GQA, bool FP8>
__global__ __launch_bounds__(NTHREADS) void attn_decode_partial_t(
    float* __restrict__ partials,
    const bf16_t* __restrict__ q,        // [B, H, 128], row stride q_tstride
    const char* __restrict__ k_cache,    // [NB, KVH, BS, 128], bf16 or fp8
    const char* __restrict__ v_cache,
    const float* __restrict__ k_scale,   // [NB, KVH, BS] per-row dequant (FP8 only)
    const float* __restrict__ v_scale,
    const int* __restrict__ block_tables,  // [B, max_blocks]
    const int* __restrict__ context_lens,  // [B]
    float scale, int num_kv_heads,
    int block_size, int max_blocks, int max_chunks, int q_tstride, int chunk_keys) {
  const int b = blockIdx.x;
  const int g_kv = blockIdx.y;
  const int chunk = blockIdx.z;
  const int L = context_lens[b];
  const int c0 = chunk * chunk_keys;
  if (c0 >= L && chunk > 0) return;     // no keys for this chunk
  const int c1 = min(L, c0 + chunk_keys);
  const int tid = threadIdx.x;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* q_lds = reinterpret_cast<float*>(smem);          // [GQA][128]
  float* s_lds = q_lds + GQA * 128;                       // [GQA][TKV]
  float* mstate = s_lds + GQA * TKV;                      // [GQA]
  float* lstate = mstate + 8;                              // [GQA]
  float* alpha_lds = lstate + 8;                           // [GQA]
  float* o_scratch = alpha_lds + 8;                        // [4][GQA][128]
  int64_t* rowoff = reinterpret_cast<int64_t*>(o_scratch + 4 * GQA * 128);  // [TKV]
  float* ks_lds = reinterpret_cast<float*>(rowoff + TKV);                   // [TKV] (FP8)
  float* vs_lds = ks_lds + TKV;                                             // [TKV] (FP8)

  for (int i = tid; i < GQA * 128; i += NTHREADS) {
    const int g = i >> 7, d = i & 127;
    const bf16_t* qp = q + (int64_t)b * q_tstride + (g_kv * GQA + g) * 128;
    q_lds[i] = bf16_to_f32(((const short*)qp)[d]) * scale;
  }
  if (tid < 8) { mstate[tid] = -INFINITY; lstate[tid] = 0.0f; }
  __syncthreads();

  float o_part[GQA][2];
#pragma unroll
  for (int g = 0; g < GQA; ++g) { o_part[g][0] = 0.f; o_part[g][1] = 0.f; }

  const int* bt = block_tables + (int64_t)b * max_blocks;

  for (int tile = c0; tile < c1; tile += TKV) {
    const int nkeys = min(TKV, c1 - tile);
    // stage per-key cache-row element offsets (shared by K and V) and, for
    // fp8 caches, the per-row dequant scales (scale row index = rowoff/128)
    for (int i = tid; i < nkeys; i += NTHREADS) {
      const int gk = tile + i;
      const int64_t off = (((int64_t)bt[gk / block_size] * num_kv_heads + g_kv) * block_size +
                           (gk % block_size)) * 128;
      rowoff[i] = off;
      if constexpr (FP8) {
        ks_lds[i] = k_scale[off >> 7];
        vs_lds[i] = v_scale[off >> 7];
      }
    }
    __syncthreads();

    // ---- scores: KLANES lanes per key, 128/KLANES bf16 per lane -----------
    {
      constexpr int ELEMS = 128 / KLANES;     // per-lane K elements
      constexpr int VECS = ELEMS / 8;         // 16-byte pieces per lane
      const int kg = tid / KLANES;
      const int kl = tid % KLANES;
      for (int key0 = 0; key0 < nkeys; key0 += NTHREADS / KLANES) {
        const int key = key0 + kg;
        if (key < nkeys) {
          float af[ELEMS];
          if constexpr (FP8) {
            const u8x8_vec* kv8 = reinterpret_cast<const u8x8_vec*>(k_cache + rowoff[key]) + kl * VECS;
#pragma unroll
            for (int c = 0; c < VECS; ++c) {
              u8x8_vec kv = kv8[c];
#pragma unroll
              for (int j = 0; j < 8; ++j) af[c * 8 + j] = fp8_to_f32(kv[j]);
            }
          } else {
            const bf16x8_vec* kv8 =
                reinterpret_cast<const bf16x8_vec*>(k_cache + 2 * rowoff[key]) + kl * VECS;
#pragma unroll
            for (int c = 0; c < VECS; ++c) {
              bf16x8_vec kv = kv8[c];
#pragma unroll
              for (int j = 0; j < 8; ++j) af[c * 8 + j] = bf16_to_f32(kv[j]);
            }
          }
#pragma unroll
          for (int g = 0; g < GQA; ++g) {
            const float* qg = q_lds + g * 128 + kl * ELEMS;
            float acc = 0.f;
#pragma unroll
            for (int j = 0; j < ELEMS; ++j) acc += af[j] * qg[j];
#pragma unroll
            for (int off = KLANES / 2; off > 0; off >>= 1) acc += __shfl_xor(acc, off);
            // fp8: fold the key row's dequant scale into the score (q already
            // carries the softmax scale; s_k is per-key, shared across g)
            if (kl == 0) s_lds[g * TKV + key] = FP8 ? acc * ks_lds[key] : acc;
          }
        }
      }
    }
    __syncthreads();

    // ---- online softmax: one WAVE per head, in-wave reductions only -------
    {
      const int wave = tid >> 6;
      const int wlane = tid & 63;
      for (int g = wave; g < GQA; g += NTHREADS / 64) {
        float lm = -INFINITY;
        for (int i = wlane; i < nkeys; i += 64) lm = fmaxf(lm, s_lds[g * TKV + i]);
        const float tile_max = wave_reduce_max(lm);
        const float m_old = mstate[g];
        const float m_new = fmaxf(m_old, tile_max);
        const float alpha = (m_old == -INFINITY) ? 0.0f : __expf(m_old - m_new);
        float ls = 0.f;
        for (int i = wlane; i < nkeys; i += 64) {
          const float p = __expf(s_lds[g * TKV + i] - m_new);
          s_lds[g * TKV + i] = p;
          ls += p;
        }
        const float tile_sum = wave_reduce_sum(ls);
        if (wlane == 0) {
          alpha_lds[g] = alpha;
          mstate[g] = m_new;
          lstate[g] = lstate[g] * alpha + tile_sum;
        }
      }
    }
    __syncthreads();

    // ---- V accumulation (key-partitioned, V read once) --------------------
    {
      const int kpart = tid >> 6;        // wave id = key partition
      const int d0 = (tid & 63) * 2;     // this thread's dim pair
#pragma unroll
      for (int g = 0; g < GQA; ++g) {
        const float alpha = alpha_lds[g];
        o_part[g][0] *= alpha;
        o_part[g][1] *= alpha;
      }
      const int nk4 = (nkeys + 3) >> 2;
      const int kbeg = kpart * nk4;
      const int kend = min(nkeys, kbeg + nk4);
      int key = kbeg;
      // 4 keys in flight: loads issued before use (hides the per-key chain)
      for (; key + 4 <= kend; key += 4) {
        uint32_t pairs[4];
#pragma unroll
        for (int u = 0; u < 4; ++u) {
          if constexpr (FP8)
            pairs[u] = *reinterpret_cast<const unsigned short*>(v_cache + rowoff[key + u] + d0);
          else
            pairs[u] = *reinterpret_cast<const uint32_t*>(v_cache + 2 * (rowoff[key + u] + d0));
        }
#pragma unroll
        for (int u = 0; u < 4; ++u) {
          const float vs = FP8 ? vs_lds[key + u] : 1.0f;  // per-row V dequant
          const float v0 = vs * (FP8 ? fp8_to_f32((unsigned char)(pairs[u] & 0xff))
                                     : bf16_to_f32((short)(pairs[u] & 0xffff)));
          const float v1 = vs * (FP8 ? fp8_to_f32((unsigned char)((pairs[u] >> 8) & 0xff))
                                     : bf16_to_f32((short)(pairs[u] >> 16)));
#pragma unroll
          for (int g = 0; g < GQA; ++g) {
            const float p = s_lds[g * TKV + key + u];  // wave-uniform: LDS broadcast
            o_part[g][0] += p * v0;
            o_part[g][1] += p * v1;
          }
        }
      }
      for (; key < kend; ++key) {
        uint32_t pair;
        if constexpr (FP8)
          pair = *reinterpret_cast<const unsigned short*>(v_cache + rowoff[key] + d0);
        else
          pair = *reinterpret_cast<const uint32_t*>(v_cache + 2 * (rowoff[key] + d0));
        const float vs = FP8 ? vs_lds[key] : 1.0f;
        const float v0 = vs * (FP8 ? fp8_to_f32((unsigned char)(pair & 0xff))
                                   : bf16_to_f32((short)(pair & 0xffff)));
        const float v1 = vs * (FP8 ? fp8_to_f32((unsigned char)((pair >> 8) & 0xff))
                                   : bf16_to_f32((short)(pair >> 16)));
#pragma unroll
        for (int g = 0; g < GQA; ++g) {
          const float p = s_lds[g * TKV + key];
          o_part[g][0] += p * v0;
          o_part[g][1] += p * v1;
        }
      }
    }
    __syncthreads();
  }

  // ---- combine the 4 key partitions, write partial (m, l, o) --------------
  {
    const int kpart = tid >> 6;
    const int d0 = (tid & 63) * 2;
#pragma unroll
    for (int g = 0; g < GQA; ++g) {
      o_scratch[((kpart * GQA + g) * 128) + d0] = o_part[g][0];
      o_scratch[((kpart * GQA + g) * 128) + d0 + 1] = o_part[g][1];
    }
  }
  __syncthreads();
  float* base = partials +
      ((((int64_t)b * num_kv_heads + g_kv) * max_chunks + chunk) * GQA) * PART_STRIDE;
  for (int i = tid; i < GQA * 128; i += NTHREADS) {
    const int g = i >> 7, d = i & 127;
    float acc = 0.f;
#pragma unroll
    for (int kp = 0; kp < 4; ++kp) acc += o_scratch[((kp * GQA + g) * 128) + d];
    base[g * PART_STRIDE + d] = acc;
  }
  if (tid < 8 && tid < GQA) {
    float* pg = base + tid * PART_STRIDE;
    pg[128] = mstate[tid];
    pg[129] = lstate[tid];
  }
}

// host-side dispatcher: picks the GQA template instantiation
extern "C" void launch_attn_decode_partial(
    float* partials, const bf16_t* q, const void* k_cache, const void* v_cache,
    const float* k_scale, const float* v_scale,
    const int* block_tables, const int* context_lens, float scale,
    int num_q_heads, int num_kv_heads, int block_size, int max_blocks,
    int max_chunks, int q_tstride, int chunk_keys, int B, int cache_fp8,
    hipStream_t stream) {
  const int gqa = num_q_heads / num_kv_heads;
  const dim3 grid(B, num_kv_heads, max_chunks);
  const size_t lds = (gqa * 128 + gqa * TKV + 24 + 4 * gqa * 128) * sizeof(float) +
                     TKV * sizeof(int64_t) + 2 * TKV * sizeof(float);
#define LAUNCH(G, F)                                                                  \
  hipLaunchKernelGGL((attn_decode_partial_t<G, F>), grid, dim3(NTHREADS), lds, stream, \
                     partials, q, (const char*)k_cache, (const char*)v_cache,          \
                     k_scale, v_scale,                                                 \
                     block_tables, context_lens, scale, num_kv_heads, block_size,      \
                     max_blocks, max_chunks, q_tstride, chunk_keys)
#define DISPATCH(G) do { if (cache_fp8) LAUNCH(G, true); else LAUNCH(G, false); } while (0)
  switch (gqa) {
    case 1: DISPATCH(1); break;
    case 2: DISPATCH(2); break;
    case 4: DISPATCH(4); break;
    case 8: DISPATCH(8); break;
    default: DISPATCH(8); break;  // guarded by the binding's gqa<=8 check
  }
#undef DISPATCH
#undef LAUNCH
}

// merge the per-chunk partials: one 64-lane wave per (b, q_head)
extern "C" __global__ void __launch_bounds__(64) attn_decode_reduce_kernel(
    bf16_t* __restrict__ out,            // [B, H, 128]
    const float* __restrict__ partials,  // [B, KVH, max_chunks, gqa, 130]
    const int* __restrict__ context_lens,
    int num_q_heads, int num_kv_heads, int max_chunks, int chunk_keys) {
  const int b = blockIdx.x;
  const int h = blockIdx.y;
  const int gqa = num_q_heads / num_kv_heads;
  const int g_kv = h / gqa;
  const int g = h % gqa;
  const int lane = threadIdx.x;
  const int nchunks = min(max_chunks, (context_lens[b] + chunk_keys - 1) / chunk_keys);

  const float* base = partials +
      ((((int64_t)b * num_kv_heads + g_kv) * max_chunks) * gqa + g) * PART_STRIDE;

  // global max over chunks
  float m = -INFINITY;
  for (int c = 0; c < nchunks; ++c) m = fmaxf(m, base[(int64_t)c * gqa * PART_STRIDE + 128]);
  float l_tot = 0.0f;
  float acc0 = 0.0f, acc1 = 0.0f;   // two dims per lane (d = lane*2)
  for (int c = 0; c < nchunks; ++c) {
    const float* pc = base + (int64_t)c * gqa * PART_STRIDE;
    const float w = __expf(pc[128] - m);
    l_tot += pc[129] * w;
    acc0 += pc[lane * 2] * w;
    acc1 += pc[lane * 2 + 1] * w;
  }
  const float inv_l = (l_tot > 0.f) ? 1.0f / l_tot : 0.0f;
  bf16_t* op = out + (((int64_t)b * num_q_heads) + h) * 128;
  const uint32_t pair = ((uint32_t)(uint16_t)f32_to_bf16(acc1 * inv_l) << 16) |
                        (uint16_t)f32_to_bf16(acc0 * inv_l);
  *reinterpret_cast<uint32_t*>((short*)op + lane * 2) = pair;
}
