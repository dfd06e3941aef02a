"""Time the grouped expert GEMM kernels at a granite-3b-moe-ish shape.
Run on a GPU box: python tools_moe_gemm_probe.py"""
import torch

from dolomite_engine_amd.ops import hip


def bench_op(fn, iters=20):
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / iters


def main():
    torch.manual_seed(0)
    # 8k tokens * top2 routing over 32 experts, h=1024, ffn 512 (glu -> 1024 out)
    for (E, N, K, T) in [(32, 1024, 1024, 16384), (8, 4096, 1024, 32768), (8, 1024, 2048, 32768)]:
        counts = torch.full((E,), T // E, dtype=torch.int64, device="cuda")
        x = (torch.randn(T, K, device="cuda") * 0.1).bfloat16()
        w = (torch.randn(E, N, K, device="cuda") * 0.02).bfloat16()
        dy = (torch.randn(T, N, device="cuda") * 0.1).bfloat16()
        y = torch.empty(T, N, dtype=torch.bfloat16, device="cuda")
        dx = torch.empty_like(x)
        dw = torch.empty_like(w)
        offsets = torch.zeros(E + 1, dtype=torch.int32, device="cuda")
        offsets[1:] = counts.cumsum(0).to(torch.int32)
        mr = int(counts.max())

        flops = 2.0 * T * N * K

        ms = bench_op(lambda: hip.check(hip.lib().dolomite_moe_gemm_fwd(
            hip.stream(), hip.ptr(x), hip.ptr(w), None, hip.ptr(y), hip.ptr(offsets), E, mr, N, K, 1), "f"))
        tf_f = flops / ms / 1e9
        ms = bench_op(lambda: hip.check(hip.lib().dolomite_moe_gemm_dgrad(
            hip.stream(), hip.ptr(dy), hip.ptr(w), hip.ptr(dx), hip.ptr(offsets), E, mr, N, K, 1), "d"))
        tf_d = flops / ms / 1e9
        ms = bench_op(lambda: hip.check(hip.lib().dolomite_moe_gemm_wgrad(
            hip.stream(), hip.ptr(dy), hip.ptr(x), hip.ptr(dw), hip.ptr(offsets), E, N, K, 1), "w"))
        tf_w = flops / ms / 1e9

        # eager loop comparison (rocBLAS per expert)
        xs = list(x.split(counts.tolist()))

        def eager():
            torch.cat([torch.nn.functional.linear(xs[i], w[i]) for i in range(E)])

        ms_e = bench_op(eager)
        tf_e = flops / ms_e / 1e9
        print(f"E={E} N={N} K={K} T={T}: fwd {tf_f:.0f} TF  dgrad {tf_d:.0f} TF  "
              f"wgrad {tf_w:.0f} TF  | eager-loop fwd {tf_e:.0f} TF")


if __name__ == "__main__":
    main()
