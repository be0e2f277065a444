// One-shot all-reduce kernel for small (latency-bound) decode-step tensors.
//
// Round-2 groundwork for SURVEY §5.8 / §2.3: on an 8-GPU MI355X node the
// xGMI fabric is fully connected (7 p2p links per GPU), so for the tens-of-KB
// tensors a TP decode step reduces ([n_streams, hidden] bf16), the
// latency-optimal schedule is ONE SHOT: every rank reads all peers' buffers
// directly over xGMI and writes the sum — no ring hops, no chunk pipeline.
//
// This kernel is the device side of that design and is testable on a single
// GPU: the peer pointers are just N buffers (same device here; IPC-mapped
// peer memory in the multi-GPU wiring, which needs an 8-GPU box to
// validate — see docs/ROADMAP.md item 1). Reduction is fp32-accumulated
// bf16, 8-element (16 B) vectorized per lane, grid-strided.
//
// Reference parity note: the reference has no collectives at all (its
// backend is a remote API); this replaces nothing and accelerates the native
// TP path only.

#include "common.h"

#define AR_MAX_PEERS 8

struct PeerPtrs {
  const bf16_t* src[AR_MAX_PEERS];
  bf16_t* dst[AR_MAX_PEERS];
};

// n8 = number of 8-element bf16 groups (caller guarantees numel % 8 == 0).
__global__ void one_shot_allreduce_kernel(PeerPtrs p, int n_peers, long n8,
                                          int write_all) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long g = i; g < n8; g += stride) {
    float acc[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
    for (int r = 0; r < n_peers; ++r) {
      bf16x8_vec v = *reinterpret_cast<const bf16x8_vec*>(p.src[r] + g * 8);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        bf16_t b = __builtin_bit_cast(bf16_t, (short)v[j]);
        acc[j] += __bfloat162float(b);
      }
    }
    bf16x8_vec out;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      out[j] = __builtin_bit_cast(short, __float2bfloat16(acc[j]));
    }
    if (write_all) {
      // single-GPU simulation / rank-0-writes-peers variant
      for (int r = 0; r < n_peers; ++r) {
        *reinterpret_cast<bf16x8_vec*>(p.dst[r] + g * 8) = out;
      }
    } else {
      // multi-GPU one-shot: each rank writes only its OWN buffer (dst[0])
      *reinterpret_cast<bf16x8_vec*>(p.dst[0] + g * 8) = out;
    }
  }
}

extern "C" void launch_one_shot_allreduce(const void* const* srcs, void* const* dsts,
                                          int n_peers, long numel, int write_all,
                                          hipStream_t stream) {
  PeerPtrs p;
  for (int r = 0; r < AR_MAX_PEERS; ++r) {
    p.src[r] = r < n_peers ? reinterpret_cast<const bf16_t*>(srcs[r]) : nullptr;
    p.dst[r] = r < n_peers ? reinterpret_cast<bf16_t*>(dsts[r]) : nullptr;
  }
  const long n8 = numel / 8;
  // latency-bound small messages: modest grid, 256 threads; cap blocks so
  // tiny tensors don't pay full-chip launch cost
  int blocks = (int)((n8 + 255) / 256);
  if (blocks > 1024) blocks = 1024;
  if (blocks < 1) blocks = 1;
  hipLaunchKernelGGL(one_shot_allreduce_kernel, dim3(blocks), dim3(256), 0, stream,
                     p, n_peers, n8, write_all);
}
