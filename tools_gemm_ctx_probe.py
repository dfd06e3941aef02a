"""Why do the fwd NT GEMMs run ~0.88 PF in-step vs 1.06-1.5 PF isolated?
Conditions per shape: warm (reused operands, = round-1 probe), cold-x
(fresh activations from a rotating pool, like in-step), heated (after a
sustained-MFMA burst, steady DVFS), mixed (alternated with a bandwidth
kernel, like the step's norm/gelu interleave). Pure analysis tool."""
import time

import torch

torch.manual_seed(0)
M = 65536
shapes = [("qkv", M, 2720, 2560), ("cfc", M, 10240, 2560), ("lmhead", M, 49152, 2560)]


def timed(fn, n=8):
    for _ in range(2):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n


print(f"{'shape':8s} {'warm':>8s} {'cold-x':>8s} {'heated':>8s} {'mixed':>8s}   (TF)")
for name, m, n, k in shapes:
    w = torch.randn(n, k, dtype=torch.bfloat16, device="cuda")
    xs = [torch.randn(m, k, dtype=torch.bfloat16, device="cuda") for _ in range(4)]
    big = torch.randn(m, 10240 if n != 10240 else 4096, dtype=torch.bfloat16, device="cuda")
    tf = 2 * m * n * k / 1e12

    t_warm = timed(lambda: torch.nn.functional.linear(xs[0], w))

    i = [0]
    def cold():
        i[0] = (i[0] + 1) % 4
        torch.nn.functional.linear(xs[i[0]], w)
    t_cold = timed(cold)

    # heated: 40 back-to-back GEMMs to settle DVFS, then measure immediately
    for _ in range(40):
        torch.nn.functional.linear(xs[0], w)
    torch.cuda.synchronize()
    t_heat = timed(lambda: torch.nn.functional.linear(xs[0], w))

    ev = [torch.cuda.Event(enable_timing=True) for _ in range(2)]
    def mixed_once():
        torch.nn.functional.gelu(big)  # ~1.3-2.6 GB bandwidth kernel
        ev[0].record()
        torch.nn.functional.linear(xs[0], w)
        ev[1].record()
    for _ in range(2):
        mixed_once()
    torch.cuda.synchronize()
    acc = 0.0
    for _ in range(8):
        mixed_once()
        torch.cuda.synchronize()
        acc += ev[0].elapsed_time(ev[1]) / 1e3
    t_mix = acc / 8

    print(f"{name:8s} {tf/t_warm:7.0f}  {tf/t_cold:7.0f}  {tf/t_heat:7.0f}  {tf/t_mix:7.0f}   ({tf:.2f} TFLOP)")
    del w, xs, big
