import time, torch
shapes = [  # the 8B prefill projections at 12288 tokens
    (12288, 6144, 4096, "qkv"),
    (12288, 4096, 4096, "o"),
    (12288, 28672, 4096, "gateup"),
    (12288, 4096, 14336, "down"),
    (12288, 128256, 4096, "lmhead-ish"),
]
for M, N, K, name in shapes:
    a = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda")
    for _ in range(3): torch.nn.functional.linear(a, w)
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(10): torch.nn.functional.linear(a, w)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / 10
    print(f"{name}: {M}x{N}x{K} {dt*1e3:.3f} ms {2*M*N*K/dt/1e12:.0f} TF/s")
