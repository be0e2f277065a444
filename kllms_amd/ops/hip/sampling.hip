// Fused sampling + logprob kernel for gfx950.
//
// One 1024-thread block per sequence row over the full vocab (grid-stride):
//   pass A: masked max (constrained-decode bitmask applied here) + argmax
//   pass B: unscaled sum-exp (for the OpenAI-style model logprob); when top-p
//           or top-k truncation is requested, also a 256-bin PER-WAVE
//           histogram of temperature-scaled probability mass + count
//           (per-wave copies kill the LDS same-bin atomic serialization that
//           made a single shared histogram ~50x slower)
//   pass C: admission threshold from the histogram suffix (top-p by mass,
//           top-k by count), then up to 3 REFINEMENT sweeps that re-histogram
//           only the boundary bin at 256x finer resolution each time —
//           threshold resolution 20/256^4 ~ 5e-9 in scaled-logit space, i.e.
//           exact top-k/top-p up to fp32 rounding of the logits themselves
//           (the old single-pass cut admitted everything
//           within one 0.078-wide bin of the true boundary)
//   pass D: Gumbel-max draw over admitted tokens — counter-based RNG
//           hash(seed, step, token): deterministic, stream-ordered, and
//           identical regardless of batch composition.
//
// temperature == 0 -> greedy (argmax + logprob only).

#include "common.h"

// sx(x) must be BIT-IDENTICAL across pass B (histogram), pass C's
// refinement rescans and pass D (admission). hipcc contracts even
// __fmul_rn/__fsub_rn pairs into an fma, and may do so at one site but not
// another — the recomputed sx of the MAX token then comes out as the
// product's rounding residual (~1e-10, either sign) instead of 0 and falls
// outside the (lo, 0] boundary interval (observed: the max missing from the
// refinement histogram on half the rows -> k+1 admitted). ONE explicit fma
// is a single IR op the compiler cannot split or re-associate, so every
// pass computes the same bits.
__device__ __forceinline__ float sx_of(float x, float inv_t, float smax) {
  return __builtin_fmaf(x, inv_t, -smax);
}

#define NT 1024
#define NW (NT / WAVE)     // 16 waves
#define SBINS 256
#define SRANGE 20.0f       // scaled-logit window below max covered by the histogram

extern "C" __global__ void __launch_bounds__(NT) sample_kernel(
    int64_t* __restrict__ out_tokens,     // [B]
    float* __restrict__ out_logprobs,     // [B]
    const float* __restrict__ logits,     // [B, V]
    const float* __restrict__ temperatures,
    const float* __restrict__ top_ps,
    const int* __restrict__ top_ks,
    const int64_t* __restrict__ seeds,
    const int64_t* __restrict__ steps,
    const uint32_t* __restrict__ mask,    // [B, ceil(V/32)] or nullptr
    int V, float* __restrict__ dbg) {   // dbg: [B, 16] or nullptr (diagnostics)
  const int b = blockIdx.x;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid / WAVE;
  const float* row = logits + (int64_t)b * V;
  const uint32_t* mrow = mask ? mask + (int64_t)b * ((V + 31) / 32) : nullptr;
  const float temp = temperatures[b];
  const float top_p = top_ps[b];
  const int top_k = top_ks[b];
  const bool truncate = (top_p < 1.0f) || (top_k > 0 && top_k < V);

  __shared__ float red[16];
  __shared__ float sh_thresh;
  __shared__ int sh_argmax;
  __shared__ unsigned long long sh_best;
  extern __shared__ __attribute__((aligned(16))) char hist_smem[];
  float* hist_mass = reinterpret_cast<float*>(hist_smem);              // [NW][SBINS]
  unsigned int* hist_cnt = reinterpret_cast<unsigned int*>(hist_mass + NW * SBINS);

  if (truncate) {
    for (int i = tid; i < NW * SBINS; i += NT) {
      hist_mass[i] = 0.0f;
      hist_cnt[i] = 0;
    }
  }
  if (tid == 0) { sh_argmax = 0; sh_best = 0; }
  __syncthreads();

  // ---- pass A: masked max + argmax ---------------------------------------
  float lmax = -INFINITY;
  int larg = -1;
  for (int i = tid; i < V; i += NT) {
    if (mrow && !((mrow[i >> 5] >> (i & 31)) & 1)) continue;
    const float x = row[i];
    if (x > lmax) { lmax = x; larg = i; }
  }
  const float gmax = block_reduce_max<16>(lmax, red);
  if (lmax == gmax && larg >= 0) sh_argmax = larg;
  __syncthreads();
  const int argmax_tok = sh_argmax;

  // ---- pass B: unscaled sumexp (+ optional scaled-mass histogram) ---------
  const float inv_t = (temp > 0.0f) ? 1.0f / temp : 0.0f;
  const float smax = gmax * inv_t;
  float lsum_unscaled = 0.0f;
  float lsum_scaled = 0.0f;
  float* my_mass = hist_mass + wid * SBINS;
  unsigned int* my_cnt = hist_cnt + wid * SBINS;
  for (int i = tid; i < V; i += NT) {
    if (mrow && !((mrow[i >> 5] >> (i & 31)) & 1)) continue;
    const float x = row[i];
    lsum_unscaled += __expf(x - gmax);
    if (truncate) {
      const float sx = sx_of(x, inv_t, smax);
      const float w = __expf(sx);
      lsum_scaled += w;
      int bin = (int)(-sx * (SBINS / SRANGE));
      bin = min(bin, SBINS - 1);
      atomicAdd(&my_mass[bin], w);
      atomicAdd(&my_cnt[bin], 1u);
    }
  }
  const float gsum_unscaled = block_reduce_sum<16>(lsum_unscaled, red);

  if (temp == 0.0f) {
    if (tid == 0) {
      out_tokens[b] = argmax_tok;
      out_logprobs[b] = row[argmax_tok] - gmax - __logf(gsum_unscaled);
    }
    return;
  }

  // ---- pass C: admission threshold ----------------------------------------
  float x_thresh = -INFINITY;
  if (truncate) {
    const float gsum_scaled = block_reduce_sum<16>(lsum_scaled, red);
    // fold the per-wave histograms: thread j < SBINS owns bin j
    if (tid < SBINS) {
      float m = 0.0f;
      unsigned int c = 0;
#pragma unroll 4
      for (int w = 0; w < NW; ++w) {
        m += hist_mass[w * SBINS + tid];
        c += hist_cnt[w * SBINS + tid];
      }
      hist_mass[tid] = m;
      hist_cnt[tid] = c;
    }
    __syncthreads();
    // boundary-interval state (sx space: 0 at the max, negative below);
    // need_* is the admission still OWED by the boundary interval after
    // everything strictly above it is admitted. MEMBERSHIP in the boundary
    // interval is tested by re-evaluating the PREVIOUS level's exact binning
    // expression (sel_*) — never by interval comparisons, which disagree
    // with the truncating multiply by an ulp at bin edges (and at sx ~ 0).
    __shared__ float sh_lo, sh_hi, sh_need_mass;
    __shared__ float sh_sel_hi, sh_sel_invw;
    __shared__ int sh_sel_cut, sh_sel_coarse;
    __shared__ unsigned int sh_need_cnt;
    __shared__ int sh_state;  // 0 = threshold at sh_lo, 1 = refine, 2 = admit all
    if (tid == 0) {
      const float need_mass = top_p < 1.0f ? top_p * gsum_scaled : INFINITY;
      const unsigned int need_cnt =
          (top_k > 0 && top_k < V) ? (unsigned)top_k : 0xffffffffu;
      float acc_mass = 0.0f;
      unsigned int acc_cnt = 0;
      int cut_bin = -1;
      for (int bn = 0; bn < SBINS; ++bn) {
        acc_mass += hist_mass[bn];
        acc_cnt += hist_cnt[bn];
        if (acc_mass >= need_mass || acc_cnt >= need_cnt) { cut_bin = bn; break; }
      }
      if (cut_bin < 0) {
        sh_state = 2;  // window never met the need: admit everything
      } else {
        const float binw = SRANGE / SBINS;
        const float before_mass = acc_mass - hist_mass[cut_bin];
        const unsigned int before_cnt = acc_cnt - hist_cnt[cut_bin];
        sh_lo = -(cut_bin + 1) * binw;
        sh_hi = -cut_bin * binw;
        sh_need_mass = need_mass - before_mass;
        sh_need_cnt = (need_cnt == 0xffffffffu) ? 0xffffffffu : need_cnt - before_cnt;
        sh_sel_coarse = 1;
        sh_sel_cut = cut_bin;
        // the whole bin is exactly owed -> no refinement needed
        sh_state = (hist_cnt[cut_bin] == sh_need_cnt) ? 0 : 1;
        if (dbg) {
          dbg[b * 16 + 0] = (float)cut_bin;
          dbg[b * 16 + 1] = (float)hist_cnt[cut_bin];
          dbg[b * 16 + 2] = (float)sh_need_cnt;
          dbg[b * 16 + 3] = (float)sh_state;
        }
      }
    }
    __syncthreads();

    // refinement: re-histogram ONLY the boundary interval at 256x finer
    // resolution; each sweep divides the threshold uncertainty by SBINS
    for (int it = 0; it < 3; ++it) {
      if (sh_state != 1) break;
      const float lo = sh_lo, hi = sh_hi;
      const float inv_w = SBINS / (hi - lo);
      for (int i = tid; i < NW * SBINS; i += NT) {
        hist_mass[i] = 0.0f;
        hist_cnt[i] = 0;
      }
      __syncthreads();
      const int sel_coarse = sh_sel_coarse;
      const int sel_cut = sh_sel_cut;
      const float sel_hi = sh_sel_hi;
      const float sel_invw = sh_sel_invw;
      for (int i = tid; i < V; i += NT) {
        if (mrow && !((mrow[i >> 5] >> (i & 31)) & 1)) continue;
        const float sx = sx_of(row[i], inv_t, smax);
        // membership = the EXACT bin the previous level counted this token
        // into equals its boundary bin (bit-identical re-evaluation)
        int pb;
        if (sel_coarse) {
          pb = min((int)(-sx * (SBINS / SRANGE)), SBINS - 1);
        } else {
          pb = min(max((int)((sel_hi - sx) * sel_invw), 0), SBINS - 1);
        }
        if (pb != sel_cut) continue;
        int bn = (int)((hi - sx) * inv_w);
        bn = min(max(bn, 0), SBINS - 1);
        atomicAdd(&my_mass[bn], __expf(sx));
        atomicAdd(&my_cnt[bn], 1u);
      }
      __syncthreads();
      if (tid < SBINS) {
        float m = 0.0f;
        unsigned int c = 0;
#pragma unroll 4
        for (int w = 0; w < NW; ++w) {
          m += hist_mass[w * SBINS + tid];
          c += hist_cnt[w * SBINS + tid];
        }
        hist_mass[tid] = m;
        hist_cnt[tid] = c;
      }
      __syncthreads();
      if (tid == 0) {
        float acc_mass = 0.0f;
        unsigned int acc_cnt = 0;
        int cut_bin = SBINS - 1;  // need is guaranteed met inside [lo, hi)
        for (int bn = 0; bn < SBINS; ++bn) {
          acc_mass += hist_mass[bn];
          acc_cnt += hist_cnt[bn];
          if (acc_mass >= sh_need_mass || acc_cnt >= sh_need_cnt) { cut_bin = bn; break; }
        }
        const float subw = (hi - lo) / SBINS;
        sh_need_mass -= acc_mass - hist_mass[cut_bin];
        if (sh_need_cnt != 0xffffffffu) sh_need_cnt -= acc_cnt - hist_cnt[cut_bin];
        sh_lo = hi - (cut_bin + 1) * subw;
        sh_hi = hi - cut_bin * subw;
        sh_sel_coarse = 0;
        sh_sel_cut = cut_bin;
        sh_sel_hi = hi;
        sh_sel_invw = inv_w;
        sh_state = (hist_cnt[cut_bin] == sh_need_cnt) ? 0 : 1;
        if (dbg) {
          dbg[b * 16 + 4 + it * 4 + 0] = (float)cut_bin;
          dbg[b * 16 + 4 + it * 4 + 1] = (float)hist_cnt[cut_bin];
          dbg[b * 16 + 4 + it * 4 + 2] = (float)sh_need_cnt;
          dbg[b * 16 + 4 + it * 4 + 3] = sh_lo;
          if (it == 0) {
            dbg[b * 16 + 12] = (float)hist_cnt[0];
            dbg[b * 16 + 13] = sx_of(row[argmax_tok], inv_t, smax);
            dbg[b * 16 + 14] = smax;
            dbg[b * 16 + 15] = __fmul_rn(row[argmax_tok], inv_t);
          }
        }
      }
      __syncthreads();
    }
    if (sh_state != 2) {
      // threshold stays in SX space: pass D recomputes each token's sx with
      // the IDENTICAL f32 expression and compares sx > lo, so the decision
      // is bit-consistent with the refinement that chose lo (converting
      // back to x-space loses the ~1e-9 refined margin to fp32 rounding)
      if (tid == 0) sh_thresh = sh_lo;
      __syncthreads();
      x_thresh = sh_thresh;
    }
  }

  // ---- pass D: Gumbel-max over admitted tokens ----------------------------
  const uint64_t base = splitmix64((uint64_t)seeds[b] * 0x9E3779B97F4A7C15ULL +
                                   (uint64_t)steps[b] * 0xBF58476D1CE4E5B9ULL);
  float best_key = -INFINITY;
  int best_tok = argmax_tok;
  for (int i = tid; i < V; i += NT) {
    if (mrow && !((mrow[i >> 5] >> (i & 31)) & 1)) continue;
    const float x = row[i];
    // sx-space admission, exclusive lower edge (matches the (lo, hi] bin
    // convention of passes B/C); x_thresh = -inf admits everything
    if (sx_of(x, inv_t, smax) <= x_thresh) continue;
    const float u = u64_to_uniform(splitmix64(base ^ (uint64_t)i));
    const float gumbel = -__logf(-__logf(u));
    const float key = x * inv_t + gumbel;
    if (key > best_key) { best_key = key; best_tok = i; }
  }
  {
    uint32_t kbits = __float_as_uint(best_key);
    kbits = (kbits & 0x80000000u) ? ~kbits : (kbits | 0x80000000u);
    const unsigned long long packed = ((unsigned long long)kbits << 32) | (uint32_t)best_tok;
    atomicMax(&sh_best, packed);
  }
  __syncthreads();
  if (tid == 0) {
    const int tok = (int)(sh_best & 0xffffffffu);
    out_tokens[b] = tok;
    out_logprobs[b] = row[tok] - gmax - __logf(gsum_unscaled);
  }
}
