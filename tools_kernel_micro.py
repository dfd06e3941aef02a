"""Standalone micro-timing for the small bandwidth kernels (preprocess,
finalize, norm bwd) against their algorithmic byte counts. Event-timed,
K iterations back-to-back, no profiler. Pure analysis tool."""
import torch

from dolomite_engine_amd.ops import functional as Fx
from dolomite_engine_amd.ops import hip

T, H, Hkv, D = 65536, 32, 1, 80
dev = "cuda"
torch.manual_seed(0)
PEAK = 6.3e3  # GB/s measured


def timed(fn, k=30):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    e0 = torch.cuda.Event(enable_timing=True)
    e1 = torch.cuda.Event(enable_timing=True)
    e0.record()
    for _ in range(k):
        fn()
    e1.record()
    torch.cuda.synchronize()
    return e0.elapsed_time(e1) / k


# --- fa_bwd_preprocess: delta[h,t] = rowsum(dO*O) ---
o = torch.randn(T, H * D, dtype=torch.bfloat16, device=dev)
do = torch.randn(T, H * D, dtype=torch.bfloat16, device=dev)
delta = torch.empty(H, T, dtype=torch.float32, device=dev)


def pre():
    hip.check(
        hip.lib().dolomite_fa_bwd_preprocess(
            hip.stream(), hip.ptr(o), hip.ptr(do), hip.ptr(delta), T, H, D, H * D, H * D, hip.BF16
        ),
        "pre",
    )


ms = timed(pre)
gb = (2 * T * H * D * 2 + H * T * 4) / 1e9
print(f"preprocess : {ms:6.3f} ms  {gb/ms*1e3:6.0f} GB/s ({gb/ms*1e3/PEAK*100:4.1f}% of peak)")

# correctness vs torch
ref = (o.view(T, H, D).float() * do.view(T, H, D).float()).sum(-1).t().contiguous()
err = (delta - ref).abs().max().item()
print(f"             max|err| vs torch fp32: {err:.3e}")

# --- fa_grad_finalize: reduce (T, HG, D) fp32 pairs into packed bf16 ---
G = H // Hkv
dk_acc = torch.randn(T, H, D, dtype=torch.float32, device=dev)
dv_acc = torch.randn(T, H, D, dtype=torch.float32, device=dev)
row_len = (H + 2 * Hkv) * D
dqkv = torch.empty(T, row_len, dtype=torch.bfloat16, device=dev)


def fin():
    hip.check(
        hip.lib().dolomite_fa_grad_finalize(
            hip.stream(), hip.ptr(dk_acc), hip.ptr(dv_acc), hip.ptr(dqkv),
            T, Hkv, D, G, row_len, H * D, D, (H + Hkv) * D, hip.BF16,
        ),
        "fin",
    )


ms = timed(fin)
gb = (2 * T * H * D * 4 + 2 * T * Hkv * D * 2) / 1e9
print(f"finalize   : {ms:6.3f} ms  {gb/ms*1e3:6.0f} GB/s ({gb/ms*1e3/PEAK*100:4.1f}% of peak)")

# --- rmsnorm bwd (fused dres) ---
Hn = 2560
x = torch.randn(T, Hn, dtype=torch.bfloat16, device=dev).requires_grad_(True)
w = (torch.randn(Hn, dtype=torch.bfloat16, device=dev) * 0.1 + 1.0).requires_grad_(True)
r = torch.randn(T, Hn, dtype=torch.bfloat16, device=dev).requires_grad_(True)
y, s = Fx.fused_rmsnorm(x, w, 1e-5, residual=r)
dy = torch.randn_like(y)
ds = torch.randn_like(s)
g = torch.autograd.grad((y, s), (x, r, w), (dy, ds), retain_graph=True)


def nb():
    torch.autograd.grad((y, s), (x, r, w), (dy, ds), retain_graph=True)


ms = timed(nb)
gb = (5 * T * Hn * 2) / 1e9  # dy, ds, s reads + dx write (+ s_hat cast path) approx
print(f"norm_bwd   : {ms:6.3f} ms  {gb/ms*1e3:6.0f} GB/s ({gb/ms*1e3/PEAK*100:4.1f}% of peak)")
