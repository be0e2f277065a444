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
// verified by mfma_selftest.hip). Wave w owns rows [32w, 32w+32). The block
// LOOPS the m-tiles of its (expert, n-tile) so each W panel streams from
// HBM once and re-reads from its XCD's L2.
//
// Operands go THROUGH LDS in full 128-byte lines (the guide's M=256 GEMM
// recipe): fragment-shaped direct loads put each 16-lane group on 32
// DIFFERENT cache lines per instruction (TA-path bound — measured 198 TF/s);
// cooperative row staging + XOR-swizzled ds_read_b128 fragment reads fix the
// access pattern. Double-buffered BK=64 stages, ONE barrier per stage, with
// the next stage's global loads issued before the current stage's MFMAs
// (T14: HBM/L2 latency hides under the matrix work).

#include "common.h"

#define MOE_BM 128
#define MOE_BN 64
#define MOE_BK 64                      // k-depth per stage (128 B per row)
#define ROW_B 128                      // bytes per staged row (64 bf16)

typedef __bf16 bf16x8_mfma __attribute__((ext_vector_type(8)));

__device__ __forceinline__ bf16x8_mfma load16(const bf16_t* p) {
  return *reinterpret_cast<const bf16x8_mfma*>(p);
}

__device__ __forceinline__ float silu(float x) {
  return x / (1.0f + __expf(-x));
}

// byte offset inside a staged tile, with the bank-conflict XOR swizzle
__device__ __forceinline__ int tile_off(int row, int byte_in_row) {
  return row * ROW_B + (byte_in_row ^ ((row & 7) << 4));
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
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wv = tid >> 6;
  const int half = lane >> 5;
  const int jc = lane & 31;

  const int m_begin = pad_offsets[e];
  const int m_end = pad_offsets[e + 1];
  const bf16_t* we = w + (long)e * (2 * (long)IN) * K;

  // staging assignment: thread t stages 64 B (4 x 16 B) of one row
  const int st_arow = tid >> 1;                 // A: 128 rows x 2 threads
  const int st_aoff = (tid & 1) * 32;           // element offset in the row
  const int st_brow = (tid & 127) >> 1;         // B: 64 rows x 2 threads
  const int st_boff = (tid & 1) * 32;
  const bool st_up = tid >= 128;                // low half: gate, high: up
  const bf16_t* st_bsrc =
      we + ((long)((st_up ? IN : 0) + n0 + st_brow)) * K + st_boff;

  // [2 buffers][A 16 KB + gate 8 KB + up 8 KB] (offset selection — pointer
  // arrays into LDS fail to compile as static initializers)
  __shared__ __attribute__((aligned(16))) char smem[2 * (MOE_BM + 2 * MOE_BN) * ROW_B];
  const int BUF_STRIDE = (MOE_BM + 2 * MOE_BN) * ROW_B;
  auto a_lds = [&](int b) -> char* { return smem + b * BUF_STRIDE; };
  auto g_lds = [&](int b) -> char* { return smem + b * BUF_STRIDE + MOE_BM * ROW_B; };
  auto u_lds = [&](int b) -> char* { return smem + b * BUF_STRIDE + (MOE_BM + MOE_BN) * ROW_B; };

  const int n_stages = K / MOE_BK;

  for (int m0 = m_begin; m0 < m_end; m0 += MOE_BM) {
    const int tok = sorted_ids[m0 + st_arow];
    const bf16_t* st_asrc = ((tok >= 0) ? x + (long)tok * K : zeros) + st_aoff;

    f32x16 accg[2] = {{}, {}};
    f32x16 accu[2] = {{}, {}};

    bf16x8_vec ra[4], rb[4];
    auto ld = [&](int stage) {
      const bf16_t* ap = st_asrc + stage * MOE_BK;
      const bf16_t* bp = st_bsrc + stage * MOE_BK;
#pragma unroll
      for (int p = 0; p < 4; ++p) {
        ra[p] = reinterpret_cast<const bf16x8_vec*>(ap)[p];
        rb[p] = reinterpret_cast<const bf16x8_vec*>(bp)[p];
      }
    };
    auto st = [&](int buf) {
      char* bt = st_up ? u_lds(buf) : g_lds(buf);
#pragma unroll
      for (int p = 0; p < 4; ++p) {
        *reinterpret_cast<bf16x8_vec*>(
            a_lds(buf) + tile_off(st_arow, st_aoff * 2 + p * 16)) = ra[p];
        *reinterpret_cast<bf16x8_vec*>(
            bt + tile_off(st_brow, st_boff * 2 + p * 16)) = rb[p];
      }
    };

    ld(0);
    st(0);
    __syncthreads();
    for (int stage = 0; stage < n_stages; ++stage) {
      if (stage + 1 < n_stages) ld(stage + 1);
      const int buf = stage & 1;
      const int arow = wv * 32 + jc;
#pragma unroll
      for (int wdw = 0; wdw < 4; ++wdw) {            // 4 k-windows of 16
        const int bib = wdw * 32 + half * 16;
        const bf16x8_mfma af = *reinterpret_cast<const bf16x8_mfma*>(
            a_lds(buf) + tile_off(arow, bib));
#pragma unroll
        for (int ct = 0; ct < 2; ++ct) {
          const int brow = ct * 32 + jc;
          const bf16x8_mfma gf = *reinterpret_cast<const bf16x8_mfma*>(
              g_lds(buf) + tile_off(brow, bib));
          const bf16x8_mfma uf = *reinterpret_cast<const bf16x8_mfma*>(
              u_lds(buf) + tile_off(brow, bib));
          accg[ct] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af, gf, accg[ct], 0, 0, 0);
          accu[ct] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af, uf, accu[ct], 0, 0, 0);
        }
      }
      if (stage + 1 < n_stages) st(1 - buf);
      __syncthreads();
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
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wv = tid >> 6;
  const int half = lane >> 5;
  const int jc = lane & 31;

  const int m_begin = pad_offsets[e];
  const int m_end = pad_offsets[e + 1];
  const bf16_t* we = w + (long)e * (long)H * IN;

  const int st_arow = tid >> 1;
  const int st_aoff = (tid & 1) * 32;
  const int st_brow = (tid & 127) >> 1;
  const int st_boff = (tid & 1) * 32;
  const bool st_b = tid < 128;                  // low half stages B
  const bf16_t* st_bsrc = we + ((long)(n0 + st_brow)) * IN + st_boff;

  __shared__ __attribute__((aligned(16))) char smem[2 * (MOE_BM + MOE_BN) * ROW_B];
  const int BUF_STRIDE = (MOE_BM + MOE_BN) * ROW_B;
  auto a_lds = [&](int b) -> char* { return smem + b * BUF_STRIDE; };
  auto b_lds = [&](int b) -> char* { return smem + b * BUF_STRIDE + MOE_BM * ROW_B; };

  const int n_stages = IN / MOE_BK;

  for (int m0 = m_begin; m0 < m_end; m0 += MOE_BM) {
    const bf16_t* st_asrc = act + (long)(m0 + st_arow) * IN + st_aoff;

    f32x16 acc[2] = {{}, {}};
    bf16x8_vec ra[4], rb[4];
    auto ld = [&](int stage) {
      const bf16_t* ap = st_asrc + stage * MOE_BK;
#pragma unroll
      for (int p = 0; p < 4; ++p) ra[p] = reinterpret_cast<const bf16x8_vec*>(ap)[p];
      if (st_b) {
        const bf16_t* bp = st_bsrc + stage * MOE_BK;
#pragma unroll
        for (int p = 0; p < 4; ++p) rb[p] = reinterpret_cast<const bf16x8_vec*>(bp)[p];
      }
    };
    auto st = [&](int buf) {
#pragma unroll
      for (int p = 0; p < 4; ++p) {
        *reinterpret_cast<bf16x8_vec*>(
            a_lds(buf) + tile_off(st_arow, st_aoff * 2 + p * 16)) = ra[p];
      }
      if (st_b) {
#pragma unroll
        for (int p = 0; p < 4; ++p) {
          *reinterpret_cast<bf16x8_vec*>(
              b_lds(buf) + tile_off(st_brow, st_boff * 2 + p * 16)) = rb[p];
        }
      }
    };

    ld(0);
    st(0);
    __syncthreads();
    for (int stage = 0; stage < n_stages; ++stage) {
      if (stage + 1 < n_stages) ld(stage + 1);
      const int buf = stage & 1;
      const int arow = wv * 32 + jc;
#pragma unroll
      for (int wdw = 0; wdw < 4; ++wdw) {
        const int bib = wdw * 32 + half * 16;
        const bf16x8_mfma af = *reinterpret_cast<const bf16x8_mfma*>(
            a_lds(buf) + tile_off(arow, bib));
#pragma unroll
        for (int ct = 0; ct < 2; ++ct) {
          const int brow = ct * 32 + jc;
          const bf16x8_mfma bf = *reinterpret_cast<const bf16x8_mfma*>(
              b_lds(buf) + tile_off(brow, bib));
          acc[ct] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af, bf, acc[ct], 0, 0, 0);
        }
      }
      if (stage + 1 < n_stages) st(1 - buf);
      __syncthreads();
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
