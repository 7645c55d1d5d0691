"""hipBLASLt bandwidth efficiency at llama-8B decode GEMM shapes:
C[M,N] = A[M,K] @ W[K,N].T-ish; W streaming dominates -> report GB/s."""
import sys, time, torch
torch.manual_seed(0)
dev = "cuda:0"
shapes = [  # (name, K, N) weights as Linear: W[N, K]
    ("qkv",     4096,  6144),
    ("o_proj",  4096,  4096),
    ("gate_up", 4096, 28672),
    ("down",   14336,  4096),
]
for M in (1, 16, 64, 128):
    tot_t = 0.0; tot_b = 0
    for name, K, N in shapes:
        x = torch.randn(M, K, dtype=torch.bfloat16, device=dev)
        w = torch.randn(N, K, dtype=torch.bfloat16, device=dev)
        f = lambda: torch.nn.functional.linear(x, w)
        for _ in range(10): f()
        torch.cuda.synchronize()
        n = 200
        t0 = time.perf_counter()
        for _ in range(n): f()
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / n
        gb = (K * N * 2 + M * K * 2 + M * N * 2) / 1e9
        print(f"M={M:<4} {name:<8} K={K:<6} N={N:<6} {dt*1e6:7.1f} us  {gb/dt:7.0f} GB/s")
        tot_t += dt; tot_b += gb
    print(f"M={M:<4} ALL(4 projections)          {tot_t*1e6:7.1f} us  {tot_b/tot_t:7.0f} GB/s  -> 32 layers = {tot_t*32*1e3:.2f} ms")
