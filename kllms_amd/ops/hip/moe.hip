// Grouped MoE expert GEMMs for gfx950 (Mixtral prefill path).
//
// replace the per-expert Python loop (8 experts x
// mask/gather/2 GEMM/index_add per layer) with TWO MFMA kernel launches:
//
//   moe_gateup_kernel: act[s, :] = silu(x[tok(s)] @ Wg[e]^T) * (x[tok(s)] @ Wu[e]^T)
//   moe_down_kernel:   y[s, :]   = act[s] @ Wd[e]^T
//
// over tokens SORTED by expert (host side: argsort + per-expert padding to
// BM rows; models/mixtral.py). The routing-weight multiply + scatter back to
// token order is one torch index_add afterwards.
//
// Kernel geometry (both kernels): one block = 256 threads = 4 waves computes
// a [BM=128, BN=64] output tile with v_mfma_f32_32x32x16_bf16 (fragment maps
// verified by mfma_selftest.hip). Wave w owns rows [32w, 32w+32); its A rows
// are private, and the W panel is shared via L2, so there is NO LDS staging:
// A fragments gather 16 B per lane straight from the token rows (padding
// rows read a zeros page — branchless pointer select, no per-element
// branches), W fragments read 16 B per lane from the panel rows. The block
// LOOPS the m-tiles of its (expert, n-tile), so each W panel is fetched from
// HBM once per block and re-read from its XCD's L2 (the panel is
// 64 x K x 2 B = 0.5 MB < 4 MB L2); x stays LLC-resident across n-tiles.
// MFMA:VMEM per 32-deep k-step per wave = 8 MFMA : 10 x 16 B loads, hidden
// by 2 blocks/CU of TLP (no __syncthreads in the whole kernel).

#include "common.h"

#define MOE_BM 128
#define MOE_BN 64

typedef __bf16 bf16x8_mfma __attribute__((ext_vector_type(8)));

__device__ __forceinline__ bf16x8_mfma load16(const bf16_t* p) {
  return *reinterpret_cast<const bf16x8_mfma*>(p);
}

__device__ __forceinline__ float silu(float x) {
  return x / (1.0f + __expf(-x));
}

// gate_up: x [T, K] gathered by sorted_ids; w [E, 2*IN, K] fused rows
// (gate at [0, IN), up at [IN, 2*IN)); act out [S, IN] bf16 in sorted space.
// pad_offsets [E+1]: sorted-space row range of expert e (multiples of BM).
extern "C" __global__ void __launch_bounds__(256) moe_gateup_kernel(
    bf16_t* __restrict__ act,            // [S, IN]
    const bf16_t* __restrict__ x,        // [T, K]
    const bf16_t* __restrict__ w,        // [E, 2*IN, K]
    const int* __restrict__ sorted_ids,  // [S] token index or -1 (pad)
    const int* __restrict__ pad_offsets, // [E+1]
    const bf16_t* __restrict__ zeros,    // [K] zero page for pad rows
    int IN, int K, int n_tiles) {
  const int e = blockIdx.x / n_tiles;
  const int n0 = (blockIdx.x % n_tiles) * MOE_BN;
  const int lane = threadIdx.x & 63;
  const int wv = threadIdx.x >> 6;         // wave id 0..3 -> row stripe
  const int half = lane >> 5;
  const int jc = lane & 31;

  const int m_begin = pad_offsets[e];
  const int m_end = pad_offsets[e + 1];
  const bf16_t* we = w + (long)e * (2 * (long)IN) * K;

  // W panel row pointers (per lane): gate rows n0+ct*32+jc, up rows IN+...
  const bf16_t* bg[2];
  const bf16_t* bu[2];
#pragma unroll
  for (int ct = 0; ct < 2; ++ct) {
    bg[ct] = we + ((long)(n0 + ct * 32 + jc)) * K;
    bu[ct] = we + ((long)(IN + n0 + ct * 32 + jc)) * K;
  }

  for (int m0 = m_begin; m0 < m_end; m0 += MOE_BM) {
    const int row_s = m0 + wv * 32 + jc;        // this lane's sorted row
    const int tok = sorted_ids[row_s];
    const bf16_t* arow = (tok >= 0) ? x + (long)tok * K : zeros;

    f32x16 accg[2] = {{}, {}};
    f32x16 accu[2] = {{}, {}};
    for (int kk = 0; kk < K; kk += 32) {
      const bf16x8_mfma a0 = load16(arow + kk + half * 8);
      const bf16x8_mfma a1 = load16(arow + kk + 16 + half * 8);
#pragma unroll
      for (int ct = 0; ct < 2; ++ct) {
        const bf16x8_mfma g0 = load16(bg[ct] + kk + half * 8);
        const bf16x8_mfma g1 = load16(bg[ct] + kk + 16 + half * 8);
        const bf16x8_mfma u0 = load16(bu[ct] + kk + half * 8);
        const bf16x8_mfma u1 = load16(bu[ct] + kk + 16 + half * 8);
        accg[ct] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, g0, accg[ct], 0, 0, 0);
        accg[ct] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, g1, accg[ct], 0, 0, 0);
        accu[ct] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, u0, accu[ct], 0, 0, 0);
        accu[ct] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, u1, accu[ct], 0, 0, 0);
      }
    }
    // epilogue: silu(gate) * up -> act rows (sorted space; pad rows land in
    // pad slots that the down kernel's own pad handling discards)
#pragma unroll
    for (int ct = 0; ct < 2; ++ct) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int i = (r & 3) + 8 * (r >> 2) + 4 * half;   // D-row map
        const long out_row = m0 + wv * 32 + i;
        act[out_row * IN + n0 + ct * 32 + jc] =
            __float2bfloat16(silu(accg[ct][r]) * accu[ct][r]);
      }
    }
  }
}

// down: y[s, :] = act[s] @ Wd[e]^T ; Wd [E, H, IN]; y [S, H] bf16 (sorted
// space, pad rows garbage-but-confined — rows are independent in MFMA).
extern "C" __global__ void __launch_bounds__(256) moe_down_kernel(
    bf16_t* __restrict__ y,              // [S, H]
    const bf16_t* __restrict__ act,      // [S, IN]
    const bf16_t* __restrict__ w,        // [E, H, IN]
    const int* __restrict__ pad_offsets, // [E+1]
    int H, int IN, int n_tiles) {
  const int e = blockIdx.x / n_tiles;
  const int n0 = (blockIdx.x % n_tiles) * MOE_BN;
  const int lane = threadIdx.x & 63;
  const int wv = threadIdx.x >> 6;
  const int half = lane >> 5;
  const int jc = lane & 31;

  const int m_begin = pad_offsets[e];
  const int m_end = pad_offsets[e + 1];
  const bf16_t* we = w + (long)e * (long)H * IN;

  const bf16_t* bw[2];
#pragma unroll
  for (int ct = 0; ct < 2; ++ct) bw[ct] = we + ((long)(n0 + ct * 32 + jc)) * IN;

  for (int m0 = m_begin; m0 < m_end; m0 += MOE_BM) {
    const long row_s = m0 + wv * 32 + jc;
    const bf16_t* arow = act + row_s * IN;

    f32x16 acc[2] = {{}, {}};
    for (int kk = 0; kk < IN; kk += 32) {
      const bf16x8_mfma a0 = load16(arow + kk + half * 8);
      const bf16x8_mfma a1 = load16(arow + kk + 16 + half * 8);
#pragma unroll
      for (int ct = 0; ct < 2; ++ct) {
        const bf16x8_mfma b0 = load16(bw[ct] + kk + half * 8);
        const bf16x8_mfma b1 = load16(bw[ct] + kk + 16 + half * 8);
        acc[ct] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, b0, acc[ct], 0, 0, 0);
        acc[ct] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, b1, acc[ct], 0, 0, 0);
      }
    }
#pragma unroll
    for (int ct = 0; ct < 2; ++ct) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int i = (r & 3) + 8 * (r >> 2) + 4 * half;
        const long out_row = m0 + wv * 32 + i;
        y[out_row * H + n0 + ct * 32 + jc] = __float2bfloat16(acc[ct][r]);
      }
    }
  }
}

extern "C" void launch_moe_gateup(void* act, const void* x, const void* w,
                                  const int* sorted_ids, const int* pad_offsets,
                                  const void* zeros, int E, int IN, int K,
                                  hipStream_t stream) {
  const int n_tiles = IN / MOE_BN;
  hipLaunchKernelGGL(moe_gateup_kernel, dim3(E * n_tiles), dim3(256), 0, stream,
                     (bf16_t*)act, (const bf16_t*)x, (const bf16_t*)w,
                     sorted_ids, pad_offsets, (const bf16_t*)zeros, IN, K, n_tiles);
}

extern "C" void launch_moe_down(void* y, const void* act, const void* w,
                                const int* pad_offsets, int E, int H, int IN,
                                hipStream_t stream) {
  const int n_tiles = H / MOE_BN;
  hipLaunchKernelGGL(moe_down_kernel, dim3(E * n_tiles), dim3(256), 0, stream,
                     (bf16_t*)y, (const bf16_t*)act, (const bf16_t*)w,
                     pad_offsets, H, IN, n_tiles);
}
