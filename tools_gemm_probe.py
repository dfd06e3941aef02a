import torch, time
torch.manual_seed(0)
M = 65536
shapes = [
    ("qkv_fwd",   M, 2720, 2560),
    ("proj_fwd",  M, 2560, 2560),
    ("cfc_fwd",   M, 10240, 2560),
    ("cproj_fwd", M, 2560, 10240),
    ("lmhead",    M, 49152, 2560),
]
def bench(fn, n=10):
    for _ in range(3): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n

print(f"{'name':10s} {'M':>6s} {'N':>6s} {'K':>6s}  {'fwd(NT)':>9s} {'dgrad(NN)':>9s} {'wgrad(TN)':>9s}  TF")
for name, m, n, k in shapes:
    x = torch.randn(m, k, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(n, k, dtype=torch.bfloat16, device="cuda")
    dy = torch.randn(m, n, dtype=torch.bfloat16, device="cuda")
    tf = 2 * m * n * k / 1e12
    t_fwd = bench(lambda: torch.nn.functional.linear(x, w))         # x @ w^T
    t_dgrad = bench(lambda: dy @ w)                                  # NN
    t_wgrad = bench(lambda: dy.t() @ x)                              # TN
    print(f"{name:10s} {m:6d} {n:6d} {k:6d}  {tf/t_fwd:7.0f}TF {tf/t_dgrad:7.0f}TF {tf/t_wgrad:7.0f}TF  (alg {tf:.1f} TFLOP)")
    del x, w, dy
