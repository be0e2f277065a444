// PyTorch bindings for the kllms_amd gfx950 kernels (kllms_amd._C).
// Thin launch shims: shape/dtype checks, grid math, current-HIP-stream launch.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

// kernels (defined in the sibling .hip TUs)
extern "C" __global__ void rmsnorm_kernel(bf16_t*, const bf16_t*, const bf16_t*, int, float);
extern "C" __global__ void fused_add_rmsnorm_kernel(bf16_t*, const bf16_t*, bf16_t*, const bf16_t*, int, float);
extern "C" __global__ void silu_mul_kernel(bf16_t*, const bf16_t*, const bf16_t*, int64_t, int, int);
extern "C" __global__ void rope_kernel(bf16_t*, bf16_t*, const int64_t*, const float*, int, int, int, int, int);
extern "C" __global__ void store_kv_kernel(const bf16_t*, const bf16_t*, bf16_t*, bf16_t*, const int64_t*, int, int, int, int, int);
extern "C" __global__ void store_kv_fp8_kernel(const bf16_t*, const bf16_t*, unsigned char*, unsigned char*, float*, float*, const int64_t*, int, int, int, int, int);
extern "C" void launch_attn_decode_partial(float*, const bf16_t*, const void*, const void*, const float*, const float*, const int*, const int*, float, int, int, int, int, int, int, int, int, int, hipStream_t);
extern "C" __global__ void attn_decode_reduce_kernel(bf16_t*, const float*, const int*, int, int, int, int);
extern "C" __global__ void attn_prefill_kernel(bf16_t*, const bf16_t*, const bf16_t*, const bf16_t*, const int*, const int*, const int*, float, int, int, int, int);  // grouped: [G], [G], [G*8]
extern "C" __global__ void sample_kernel(int64_t*, float*, const float*, const float*, const float*, const int*, const int64_t*, const int64_t*, const uint32_t*, int, float*);
extern "C" __global__ void mfma_selftest_kernel(float*, const bf16_t*, const bf16_t*);

#define CHECK_BF16_CONTIG(t) \
  TORCH_CHECK((t).is_cuda() && (t).is_contiguous() && (t).scalar_type() == at::kBFloat16, #t " must be contiguous bf16 on GPU")

// [T, H, D] activation views: head/dim dims packed, token stride free (so the
// fused-QKV splits need no .contiguous() copies)
#define CHECK_KV_CACHE(t) \
  TORCH_CHECK((t).is_cuda() && (t).is_contiguous() && \
              ((t).scalar_type() == at::kBFloat16 || (t).scalar_type() == at::kFloat8_e4m3fn), \
              #t " must be a contiguous bf16 or fp8-e4m3 KV cache on GPU")

#define CHECK_BF16_ROWS(t) \
  TORCH_CHECK((t).is_cuda() && (t).scalar_type() == at::kBFloat16 && (t).dim() == 3 && \
              (t).stride(2) == 1 && (t).stride(1) == (t).size(2), #t " must be a bf16 [T,H,D] row view")

static inline hipStream_t cur_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

static bf16_t* bf(torch::Tensor& t) { return reinterpret_cast<bf16_t*>(t.data_ptr()); }
static const bf16_t* cbf(const torch::Tensor& t) { return reinterpret_cast<const bf16_t*>(t.data_ptr()); }

void rmsnorm(torch::Tensor out, torch::Tensor in, torch::Tensor weight, double eps) {
  CHECK_BF16_CONTIG(out); CHECK_BF16_CONTIG(in); CHECK_BF16_CONTIG(weight);
  const int n = in.size(-1);
  TORCH_CHECK(n % 8 == 0, "hidden size must be a multiple of 8");
  const int rows = in.numel() / n;
  hipLaunchKernelGGL(rmsnorm_kernel, dim3(rows), dim3(256), 0, cur_stream(),
                     bf(out), cbf(in), cbf(weight), n, (float)eps);
}

void fused_add_rmsnorm(torch::Tensor out, torch::Tensor x, torch::Tensor residual,
                       torch::Tensor weight, double eps) {
  CHECK_BF16_CONTIG(out); CHECK_BF16_CONTIG(x); CHECK_BF16_CONTIG(residual); CHECK_BF16_CONTIG(weight);
  const int n = x.size(-1);
  TORCH_CHECK(n % 8 == 0, "hidden size must be a multiple of 8");
  const int rows = x.numel() / n;
  hipLaunchKernelGGL(fused_add_rmsnorm_kernel, dim3(rows), dim3(256), 0, cur_stream(),
                     bf(out), cbf(x), bf(residual), cbf(weight), n, (float)eps);
}

void silu_mul(torch::Tensor out, torch::Tensor gate, torch::Tensor up) {
  CHECK_BF16_CONTIG(out);
  TORCH_CHECK(gate.is_cuda() && gate.scalar_type() == at::kBFloat16 && gate.dim() == 2 &&
              gate.stride(1) == 1, "gate must be a bf16 [T, I] row view");
  TORCH_CHECK(up.sizes() == gate.sizes() && up.stride(0) == gate.stride(0) && up.stride(1) == 1);
  const int64_t ncols = gate.size(1);
  TORCH_CHECK(ncols % 8 == 0 && gate.stride(0) % 8 == 0);
  const int64_t nvec = gate.numel() / 8;
  const int grid = (int)std::min<int64_t>((nvec + 255) / 256, 2048);
  hipLaunchKernelGGL(silu_mul_kernel, dim3(grid), dim3(256), 0, cur_stream(),
                     bf(out), cbf(gate), cbf(up), nvec, (int)(ncols / 8), (int)(gate.stride(0) / 8));
}

void rope_inplace(torch::Tensor q, torch::Tensor k, torch::Tensor positions, torch::Tensor cos_sin) {
  CHECK_BF16_ROWS(q); CHECK_BF16_ROWS(k);
  TORCH_CHECK(positions.scalar_type() == at::kLong && cos_sin.scalar_type() == at::kFloat);
  const int T = q.size(0), H = q.size(1), D = q.size(2), KVH = k.size(1);
  hipLaunchKernelGGL(rope_kernel, dim3(T, H + KVH), dim3(D / 2), 0, cur_stream(),
                     bf(q), bf(k), positions.data_ptr<int64_t>(), cos_sin.data_ptr<float>(),
                     H, KVH, D, (int)q.stride(0), (int)k.stride(0));
}

void store_kv(torch::Tensor k, torch::Tensor v, torch::Tensor k_cache, torch::Tensor v_cache,
              torch::Tensor slot_mapping, torch::Tensor k_scale, torch::Tensor v_scale) {
  CHECK_BF16_ROWS(k); CHECK_BF16_ROWS(v); CHECK_KV_CACHE(k_cache); CHECK_KV_CACHE(v_cache);
  TORCH_CHECK(k.stride(0) == v.stride(0), "k/v must share the token stride");
  const int T = k.size(0), KVH = k.size(1), D = k.size(2);
  const int BS = k_cache.size(2);
  if (k_cache.scalar_type() == at::kFloat8_e4m3fn) {
    // per-row quantization: one wave per (token, head) row, 4 waves/block
    TORCH_CHECK(k_scale.numel() == k_cache.size(0) * KVH * BS && k_scale.is_contiguous() &&
                    k_scale.scalar_type() == at::kFloat && v_scale.numel() == k_scale.numel(),
                "fp8 store_kv requires [NB, KVH, BS] fp32 k_scale/v_scale");
    const int64_t rows = (int64_t)T * KVH;
    const int grid = (int)std::min<int64_t>((rows + 3) / 4, 2048);
    hipLaunchKernelGGL(store_kv_fp8_kernel, dim3(grid), dim3(256), 0, cur_stream(),
                       cbf(k), cbf(v),
                       reinterpret_cast<unsigned char*>(k_cache.data_ptr()),
                       reinterpret_cast<unsigned char*>(v_cache.data_ptr()),
                       k_scale.data_ptr<float>(), v_scale.data_ptr<float>(),
                       slot_mapping.data_ptr<int64_t>(), T, KVH, D, BS, (int)k.stride(0));
  } else {
    const int64_t total = (int64_t)T * KVH * (D / 8);
    const int grid = (int)std::min<int64_t>((total + 255) / 256, 2048);
    hipLaunchKernelGGL(store_kv_kernel, dim3(grid), dim3(256), 0, cur_stream(),
                       cbf(k), cbf(v), bf(k_cache), bf(v_cache),
                       slot_mapping.data_ptr<int64_t>(), T, KVH, D, BS, (int)k.stride(0));
  }
}

void attn_decode_paged(torch::Tensor out, torch::Tensor q, torch::Tensor k_cache,
                       torch::Tensor v_cache, torch::Tensor block_tables,
                       torch::Tensor context_lens, double scale,
                       torch::Tensor k_scale, torch::Tensor v_scale) {
  CHECK_BF16_CONTIG(out); CHECK_BF16_ROWS(q); CHECK_KV_CACHE(k_cache); CHECK_KV_CACHE(v_cache);
  const int B = q.size(0), H = q.size(1), D = q.size(2);
  const int KVH = k_cache.size(1), BS = k_cache.size(2);
  TORCH_CHECK(D == 128, "attn_decode: head_dim must be 128");
  TORCH_CHECK(H % KVH == 0, "attn_decode: H must be a multiple of KVH");
  TORCH_CHECK(H / KVH == 1 || H / KVH == 2 || H / KVH == 4 || H / KVH == 8,
              "attn_decode: GQA group must be 1, 2, 4 or 8");
  const int gqa = H / KVH;
  const int max_blocks = block_tables.size(1);
  // split-K granule: fixed 256 keys measured best across B=40..240 and
  // ctx 576..2048 (adaptive coarser chunks trade combine traffic for lost
  // block-level parallelism and net 0..-5%; chunk size stays a runtime arg)
  const int max_ctx = max_blocks * BS;
  const int CHUNK_KEYS = ATTN_DECODE_TKV;  // 256
  const int max_chunks = std::max(1, (max_ctx + CHUNK_KEYS - 1) / CHUNK_KEYS);
  auto partials = torch::empty({(int64_t)B * KVH * max_chunks * gqa * 130},
                               torch::dtype(torch::kFloat).device(q.device()));
  const int cache_fp8 = k_cache.scalar_type() == at::kFloat8_e4m3fn ? 1 : 0;
  if (cache_fp8) {
    TORCH_CHECK(k_scale.numel() == k_cache.size(0) * KVH * BS && k_scale.is_contiguous() &&
                    k_scale.scalar_type() == at::kFloat && v_scale.numel() == k_scale.numel(),
                "fp8 attn_decode requires [NB, KVH, BS] fp32 k_scale/v_scale");
  }
  launch_attn_decode_partial(
      partials.data_ptr<float>(), cbf(q), k_cache.data_ptr(), v_cache.data_ptr(),
      cache_fp8 ? k_scale.data_ptr<float>() : nullptr,
      cache_fp8 ? v_scale.data_ptr<float>() : nullptr,
      block_tables.data_ptr<int>(), context_lens.data_ptr<int>(),
      (float)scale, H, KVH, BS, max_blocks, max_chunks, (int)q.stride(0), CHUNK_KEYS, B,
      cache_fp8, cur_stream());
  hipLaunchKernelGGL(attn_decode_reduce_kernel, dim3(B, H), dim3(64), 0, cur_stream(),
                     bf(out), partials.data_ptr<float>(), context_lens.data_ptr<int>(),
                     H, KVH, max_chunks, CHUNK_KEYS);
}

void attn_prefill(torch::Tensor out, torch::Tensor q, torch::Tensor k, torch::Tensor v,
                  torch::Tensor grp_seq_start, torch::Tensor grp_qpos0,
                  torch::Tensor grp_seqlen, double scale) {
  CHECK_BF16_CONTIG(out); CHECK_BF16_ROWS(q); CHECK_BF16_ROWS(k); CHECK_BF16_ROWS(v);
  const int H = q.size(1), D = q.size(2), KVH = k.size(1);
  TORCH_CHECK(D == 128, "attn_prefill: head_dim must be 128");
  TORCH_CHECK(k.stride(0) == v.stride(0), "k/v must share the token stride");
  const int ngroups = grp_seq_start.size(0);
  TORCH_CHECK(grp_qpos0.size(0) == ngroups * 8, "grp_qpos0 must be [G*8]");
  hipLaunchKernelGGL(attn_prefill_kernel, dim3(ngroups, H), dim3(512), 0, cur_stream(),
                     bf(out), cbf(q), cbf(k), cbf(v),
                     grp_seq_start.data_ptr<int>(), grp_seqlen.data_ptr<int>(),
                     grp_qpos0.data_ptr<int>(), (float)scale, H, KVH,
                     (int)q.stride(0), (int)k.stride(0));
}

void sample(torch::Tensor tokens, torch::Tensor logprobs, torch::Tensor logits,
            torch::Tensor temperatures, torch::Tensor top_ps, torch::Tensor top_ks,
            torch::Tensor seeds, torch::Tensor steps, torch::Tensor mask,
            torch::Tensor dbg) {
  TORCH_CHECK(logits.is_cuda() && logits.is_contiguous() && logits.scalar_type() == at::kFloat);
  const int B = logits.size(0), V = logits.size(1);
  const uint32_t* mptr = mask.numel() > 0
      ? reinterpret_cast<const uint32_t*>(mask.data_ptr<int>()) : nullptr;
  const size_t lds = 16 * 256 * (sizeof(float) + sizeof(unsigned int));  // per-wave histograms
  hipLaunchKernelGGL(sample_kernel, dim3(B), dim3(1024), lds, cur_stream(),
                     tokens.data_ptr<int64_t>(), logprobs.data_ptr<float>(),
                     logits.data_ptr<float>(), temperatures.data_ptr<float>(),
                     top_ps.data_ptr<float>(), top_ks.data_ptr<int>(),
                     seeds.data_ptr<int64_t>(), steps.data_ptr<int64_t>(), mptr, V,
                     dbg.numel() ? dbg.data_ptr<float>() : nullptr);
}

torch::Tensor mfma_selftest(torch::Tensor A, torch::Tensor B) {
  // D[32,32] = A[32,16] x B[16,32], verifying the fragment maps the attention
  // kernel assumes for v_mfma_f32_32x32x16_bf16.
  CHECK_BF16_CONTIG(A); CHECK_BF16_CONTIG(B);
  auto D = torch::empty({32, 32}, torch::dtype(torch::kFloat).device(A.device()));
  hipLaunchKernelGGL(mfma_selftest_kernel, dim3(1), dim3(64), 0, cur_stream(),
                     D.data_ptr<float>(), cbf(A), cbf(B));
  return D;
}

extern "C" void launch_one_shot_allreduce(const void* const* srcs, void* const* dsts,
                                          int n_peers, long numel, int write_all,
                                          hipStream_t stream);

void one_shot_allreduce(std::vector<torch::Tensor> bufs) {
  // In-place sum across up to 8 same-shaped bf16 buffers (single-GPU
  // simulation of the one-shot xGMI all-reduce: every "peer" buffer ends
  // holding the fp32-accumulated sum). Multi-GPU wiring (IPC-mapped peer
  // pointers, write_all=0) is round-2 work — docs/ROADMAP.md item 1.
  TORCH_CHECK(!bufs.empty() && bufs.size() <= 8, "1..8 peer buffers");
  const auto numel = bufs[0].numel();
  TORCH_CHECK(numel % 8 == 0, "numel must be a multiple of 8 (16B vector path)");
  const void* srcs[8];
  void* dsts[8];
  for (size_t r = 0; r < bufs.size(); ++r) {
    CHECK_BF16_CONTIG(bufs[r]);
    TORCH_CHECK(bufs[r].numel() == numel, "peer buffers must match in size");
    srcs[r] = bufs[r].data_ptr();
    dsts[r] = bufs[r].data_ptr();
  }
  launch_one_shot_allreduce(srcs, dsts, (int)bufs.size(), (long)numel, /*write_all=*/1,
                            cur_stream());
}

// --- grouped MoE expert GEMMs (moe.hip) --------------------------------------
extern "C" void launch_moe_gateup(void* act, const void* x, const void* w,
                                  const int* sorted_ids, const int* pad_offsets,
                                  const void* zeros, int E, int IN, int K,
                                  hipStream_t stream);
extern "C" void launch_moe_down(void* y, const void* act, const void* w,
                                const int* pad_offsets, int E, int H, int IN,
                                hipStream_t stream);

void moe_gateup(torch::Tensor act, torch::Tensor x, torch::Tensor w,
                torch::Tensor sorted_ids, torch::Tensor pad_offsets, torch::Tensor zeros) {
  CHECK_BF16_CONTIG(act); CHECK_BF16_CONTIG(x); CHECK_BF16_CONTIG(w); CHECK_BF16_CONTIG(zeros);
  TORCH_CHECK(sorted_ids.scalar_type() == at::kInt && pad_offsets.scalar_type() == at::kInt,
              "routing tensors must be int32");
  const int E = w.size(0);
  const int IN = w.size(1) / 2;
  const int K = w.size(2);
  TORCH_CHECK(IN % 64 == 0 && K % 64 == 0, "IN % 64, K % 64 required");
  TORCH_CHECK(act.size(1) == IN && x.size(1) == K && zeros.numel() >= K);
  TORCH_CHECK(act.size(0) % 128 == 0, "sorted rows must be padded to 128");
  TORCH_CHECK(pad_offsets.numel() == E + 1);
  launch_moe_gateup(act.data_ptr(), x.data_ptr(), w.data_ptr(),
                    sorted_ids.data_ptr<int>(), pad_offsets.data_ptr<int>(),
                    zeros.data_ptr(), E, IN, K, cur_stream());
}

void moe_down(torch::Tensor y, torch::Tensor act, torch::Tensor w, torch::Tensor pad_offsets) {
  CHECK_BF16_CONTIG(y); CHECK_BF16_CONTIG(act); CHECK_BF16_CONTIG(w);
  const int E = w.size(0);
  const int H = w.size(1);
  const int IN = w.size(2);
  TORCH_CHECK(H % 64 == 0 && IN % 64 == 0, "H % 64, IN % 64 required");
  TORCH_CHECK(y.size(1) == H && act.size(1) == IN);
  TORCH_CHECK(pad_offsets.numel() == E + 1);
  launch_moe_down(y.data_ptr(), act.data_ptr(), w.data_ptr(),
                  pad_offsets.data_ptr<int>(), E, H, IN, cur_stream());
}

// --- batched Levenshtein (levenshtein.hip) -----------------------------------
extern "C" void launch_levenshtein(int* out, const unsigned char* chars, const int* lens,
                                   const int* pi, const int* pj, long n_pairs,
                                   hipStream_t stream);

torch::Tensor levenshtein_pairs(torch::Tensor chars, torch::Tensor lens,
                                torch::Tensor pair_i, torch::Tensor pair_j) {
  TORCH_CHECK(chars.is_cuda() && chars.scalar_type() == at::kByte && chars.size(1) == 64 &&
              chars.is_contiguous(), "chars must be contiguous uint8 [N, 64] on GPU");
  TORCH_CHECK(lens.scalar_type() == at::kInt && pair_i.scalar_type() == at::kInt &&
              pair_j.scalar_type() == at::kInt);
  const long P = pair_i.numel();
  auto out = torch::empty({P}, torch::dtype(torch::kInt).device(chars.device()));
  launch_levenshtein(out.data_ptr<int>(),
                     chars.data_ptr<unsigned char>(), lens.data_ptr<int>(),
                     pair_i.data_ptr<int>(), pair_j.data_ptr<int>(), P, cur_stream());
  return out;
}

// --- hipIpc one-shot all-reduce (allreduce.hip) ------------------------------
extern "C" void* ipc_ar_create(int rank, int world, size_t max_bytes, unsigned char* handles_out);
extern "C" int ipc_ar_connect(void* ctx, const unsigned char* all_handles);
extern "C" int ipc_ar_run(void* ctx, const void* inp, void* out, long numel, hipStream_t stream);
extern "C" void ipc_ar_destroy(void* ctx);

std::pair<int64_t, py::bytes> ipc_allreduce_create(int rank, int world, int64_t max_bytes) {
  unsigned char handles[128];
  void* ctx = ipc_ar_create(rank, world, (size_t)max_bytes, handles);
  TORCH_CHECK(ctx != nullptr, "ipc_ar_create failed (see stderr)");
  return {reinterpret_cast<int64_t>(ctx), py::bytes(reinterpret_cast<char*>(handles), 128)};
}

void ipc_allreduce_connect(int64_t ctx, py::bytes all_handles) {
  std::string h = all_handles;
  TORCH_CHECK(h.size() % 128 == 0, "handle blob must be world*128 bytes");
  int rc = ipc_ar_connect(reinterpret_cast<void*>(ctx),
                          reinterpret_cast<const unsigned char*>(h.data()));
  TORCH_CHECK(rc == 0, "ipc_ar_connect failed (see stderr)");
}

void ipc_allreduce_run(int64_t ctx, torch::Tensor inp, torch::Tensor out) {
  CHECK_BF16_CONTIG(inp);
  CHECK_BF16_CONTIG(out);
  TORCH_CHECK(inp.numel() == out.numel(), "in/out size mismatch");
  int rc = ipc_ar_run(reinterpret_cast<void*>(ctx), inp.data_ptr(), out.data_ptr(),
                      (long)inp.numel(), cur_stream());
  TORCH_CHECK(rc == 0, "ipc_ar_run failed rc=", rc);
}

void ipc_allreduce_destroy(int64_t ctx) {
  ipc_ar_destroy(reinterpret_cast<void*>(ctx));
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("levenshtein_pairs", &levenshtein_pairs);
  m.def("moe_gateup", &moe_gateup);
  m.def("moe_down", &moe_down);
  m.def("ipc_allreduce_create", &ipc_allreduce_create);
  m.def("ipc_allreduce_connect", &ipc_allreduce_connect);
  m.def("ipc_allreduce_run", &ipc_allreduce_run);
  m.def("ipc_allreduce_destroy", &ipc_allreduce_destroy);
  m.def("rmsnorm", &rmsnorm);
  m.def("fused_add_rmsnorm", &fused_add_rmsnorm);
  m.def("silu_mul", &silu_mul);
  m.def("rope_inplace", &rope_inplace);
  m.def("store_kv", &store_kv);
  m.def("attn_decode_paged", &attn_decode_paged);
  m.def("attn_prefill", &attn_prefill);
  m.def("sample", &sample);
  m.def("mfma_selftest", &mfma_selftest);
  m.def("one_shot_allreduce", &one_shot_allreduce);
}
