import time
import torch
torch.manual_seed(0)
M = 65536
shapes = [("qkv", M, 2720, 2560), ("proj", M, 2560, 2560), ("cfc", M, 10240, 2560), ("cproj", M, 2560, 10240)]

def timed(fn, n=8):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n

print(f"{'shape':6s} {'nobias':>8s} {'bias':>8s} {'gemm+add':>9s}  (TF)")
for name, m, n, k in shapes:
    x = torch.randn(m, k, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(n, k, dtype=torch.bfloat16, device="cuda")
    b = torch.randn(n, dtype=torch.bfloat16, device="cuda")
    tf = 2 * m * n * k / 1e12
    t0 = timed(lambda: torch.nn.functional.linear(x, w))
    t1 = timed(lambda: torch.nn.functional.linear(x, w, b))
    t2 = timed(lambda: torch.nn.functional.linear(x, w).add_(b))
    print(f"{name:6s} {tf/t0:7.0f}  {tf/t1:7.0f}  {tf/t2:8.0f}   ({tf:.2f})")
    del x, w, b
