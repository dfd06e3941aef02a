"""Replicate the per-layer fwd GEMM chain to localize why c_attn/c_proj
run ~400 TF in-step vs >1 PF isolated. Variants:
  chain  — norm-mimic -> qkv -> attn-mimic -> proj -> norm -> cfc -> gelu -> mlpproj,
           fresh output allocations per call (exactly the model's fwd shape cycle)
  cycle  — the 4 GEMMs back-to-back, no elementwise kernels in between
  prealloc — cycle but with out= preallocated buffers (no allocator traffic)
  autograd — chain under autograd with the graph kept alive (real activation
             footprint, ~150 GB retained, autograd dispatch overhead)
  ballast  — cycle with ~120 GB of dead retained tensors (memory footprint
             alone, no autograd)
Pure analysis tool.
"""
import time

import torch

torch.manual_seed(0)
M, H, FF, QKV = 65536, 2560, 10240, 2720
L = 32
dt = torch.bfloat16
dev = "cuda"

w_qkv = torch.randn(QKV, H, dtype=dt, device=dev) * 0.02
b_qkv = torch.randn(QKV, dtype=dt, device=dev)
w_proj = torch.randn(H, H, dtype=dt, device=dev) * 0.02
b_proj = torch.randn(H, dtype=dt, device=dev)
w_cfc = torch.randn(FF, H, dtype=dt, device=dev) * 0.02
b_cfc = torch.randn(FF, dtype=dt, device=dev)
w_mproj = torch.randn(H, FF, dtype=dt, device=dev) * 0.02
b_mproj = torch.randn(H, dtype=dt, device=dev)

x = torch.randn(M, H, dtype=dt, device=dev)
tfs = [2 * M * QKV * H / 1e12, 2 * M * H * H / 1e12, 2 * M * FF * H / 1e12, 2 * M * H * FF / 1e12]
names = ["qkv", "proj", "cfc", "mproj"]


def run(variant: str, passes: int = 3):
    # one event pair per (layer, position); sync only at pass end so the
    # device-side back-to-back structure matches the real step
    pairs = [[(torch.cuda.Event(enable_timing=True), torch.cuda.Event(enable_timing=True))
              for _ in range(4)] for _ in range(L)]
    acc = [0.0] * 4
    n = 0
    # preallocated outputs for the no-alloc variant
    o = [torch.empty(M, QKV, dtype=dt, device=dev), torch.empty(M, H, dtype=dt, device=dev),
         torch.empty(M, FF, dtype=dt, device=dev), torch.empty(M, H, dtype=dt, device=dev)]
    ballast = []
    if variant == "ballast":
        for _ in range(60):
            ballast.append(torch.empty(1 << 30, dtype=dt, device=dev))  # 60 x 2 GiB
    for p in range(passes + 1):  # pass 0 = warmup, untimed
        h = x.requires_grad_(False)
        if variant == "autograd":
            h = x.clone().requires_grad_(True)
        keep = []
        for li in range(L):
            evs = pairs[li]
            if variant in ("chain", "autograd"):
                h = h * torch.rsqrt(h.float().pow(2).mean(-1, keepdim=True) + 1e-5).to(dt)  # norm mimic
                evs[0][0].record(); qkv = torch.nn.functional.linear(h, w_qkv, b_qkv); evs[0][1].record()
                a = qkv[:, :H].contiguous()  # attn-output mimic (copy kernel)
                evs[1][0].record(); h2 = torch.nn.functional.linear(a, w_proj, b_proj); evs[1][1].record()
                h = h + h2
                g = h * torch.rsqrt(h.float().pow(2).mean(-1, keepdim=True) + 1e-5).to(dt)
                evs[2][0].record(); u = torch.nn.functional.linear(g, w_cfc, b_cfc); evs[2][1].record()
                u = torch.nn.functional.gelu(u)
                evs[3][0].record(); h2 = torch.nn.functional.linear(u, w_mproj, b_mproj); evs[3][1].record()
                h = h + h2
                if variant == "autograd":
                    keep.append((qkv, u, h2))  # hold the graph + activations
            elif variant in ("cycle", "ballast"):
                evs[0][0].record(); qkv = torch.nn.functional.linear(h, w_qkv, b_qkv); evs[0][1].record()
                a = qkv[:, :H].contiguous()
                evs[1][0].record(); a = torch.nn.functional.linear(a, w_proj, b_proj); evs[1][1].record()
                evs[2][0].record(); u = torch.nn.functional.linear(a, w_cfc, b_cfc); evs[2][1].record()
                evs[3][0].record(); h = torch.nn.functional.linear(u[:, :FF], w_mproj, b_mproj); evs[3][1].record()
            else:  # prealloc: mm into fixed buffers, no bias, no allocs
                evs[0][0].record(); torch.matmul(h, w_qkv.t(), out=o[0]); evs[0][1].record()
                a = o[0][:, :H]
                evs[1][0].record(); torch.matmul(a.contiguous(), w_proj.t(), out=o[1]); evs[1][1].record()
                evs[2][0].record(); torch.matmul(o[1], w_cfc.t(), out=o[2]); evs[2][1].record()
                evs[3][0].record(); torch.matmul(o[2], w_mproj.t(), out=o[3]); evs[3][1].record()
                h = o[3]
        torch.cuda.synchronize()
        if p > 0:
            for li in range(L):
                for i in range(4):
                    acc[i] += pairs[li][i][0].elapsed_time(pairs[li][i][1]) / 1e3
            n += L
    print(f"{variant:9s} " + "  ".join(f"{names[i]}={tfs[i]*n/acc[i]:5.0f}TF({acc[i]/n*1e3:5.2f}ms)" for i in range(4)))


import sys

variants = sys.argv[1:] or ["chain", "cycle", "prealloc", "autograd", "ballast"]
for v in variants:
    run(v)
    torch.cuda.empty_cache()
