"""Event-timed fwd GEMMs inside the REAL bench step (flagship 3B, B=16),
independent of rocprof attribution: CUDA-event pairs around every Linear
module's forward, synced at step end. Pure analysis tool."""
import argparse
from collections import defaultdict

import torch

import bench as B


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="3b")
    ap.add_argument("--micro-batch", type=int, default=B.MICRO_BATCH)
    ap.add_argument("--seq-len", type=int, default=B.SEQ_LEN)
    ap.add_argument("--steps", type=int, default=2)
    args = ap.parse_args()

    dev = torch.device("cuda")
    torch.manual_seed(0)
    wrapper, engine = B.build(args, dev)
    from dolomite_engine_amd.optimization import LRScheduler

    sched = LRScheduler(1e-5, 0, 0, None, 10**9, "constant", 0.1)

    rec = defaultdict(list)  # short name -> [(ev_start, ev_end)]

    def mk(nm):
        def pre(mod, inp):
            e = torch.cuda.Event(enable_timing=True)
            e.record()
            mod._t0 = e

        def post(mod, inp, out):
            e = torch.cuda.Event(enable_timing=True)
            e.record()
            rec[nm].append((mod._t0, e))
            # bracket this call's backward node (dgrad+wgrad for Linears)
            if getattr(out, "grad_fn", None) is not None:
                node, cell = out.grad_fn, [None]

                def bpre(_g, cell=cell):
                    cell[0] = torch.cuda.Event(enable_timing=True)
                    cell[0].record()

                def bpost(_gi, _go, cell=cell):
                    e1 = torch.cuda.Event(enable_timing=True)
                    e1.record()
                    rec[nm + "_bwd"].append((cell[0], e1))

                node.register_prehook(bpre)
                node.register_hook(bpost)

        return pre, post

    hooks = []
    # whole-forward device interval (root module): arbitrates between the
    # rocprof per-kernel durations and the per-linear event intervals
    fpre, fpost = mk("WHOLE_FWD")
    hooks.append(wrapper.model.register_forward_pre_hook(fpre))
    hooks.append(wrapper.model.register_forward_hook(fpost))

    from dolomite_engine_amd.hf_models.modeling import Attention, GPTDolomiteBlock

    def mkb(nm):
        def bpre(mod, gout):
            e = torch.cuda.Event(enable_timing=True)
            e.record()
            mod._bt0 = e

        def bpost(mod, gin, gout):
            e = torch.cuda.Event(enable_timing=True)
            e.record()
            rec[nm].append((mod._bt0, e))

        return bpre, bpost

    # The padding-free path calls forward_padding_free directly (no module
    # __call__), so bracket the attention core at the autograd.Function level.
    import dolomite_engine_amd.ops.functional as F

    def wrap_fn(cls, nm):
        fwd0, bwd0 = cls.forward, cls.backward

        def fwd(ctx, *a, **k):
            e0 = torch.cuda.Event(enable_timing=True)
            e0.record()
            out = fwd0(ctx, *a, **k)
            e1 = torch.cuda.Event(enable_timing=True)
            e1.record()
            rec[nm + "_fwd"].append((e0, e1))
            return out

        def bwd(ctx, *g):
            e0 = torch.cuda.Event(enable_timing=True)
            e0.record()
            out = bwd0(ctx, *g)
            e1 = torch.cuda.Event(enable_timing=True)
            e1.record()
            rec[nm + "_bwd"].append((e0, e1))
            return out

        cls.forward, cls.backward = staticmethod(fwd), staticmethod(bwd)

    for cls, nm in [(F.VarlenAttention, "ATTN"), (F.RoPEPackedQKV, "ROPE"),
                    (F.FusedRMSNorm, "NORM"), (F.FusedCrossEntropy, "CE")]:
        wrap_fn(cls, nm)
    for name, mod in wrapper.model.named_modules():
        if isinstance(mod, torch.nn.Linear):
            short = ".".join(name.split(".")[-2:])  # attn.c_proj vs mlp.c_proj
            pre, post = mk(short)
            hooks.append(mod.register_forward_pre_hook(pre))
            hooks.append(mod.register_forward_hook(post))

    vocab = wrapper.model.config.vocab_size

    def step(i):
        # train_step with region brackets (same op order as train_utils.train_step)
        def ev():
            e = torch.cuda.Event(enable_timing=True)
            e.record()
            return e

        engine.zero_grad()
        engine.set_sync(True)
        batch = B.synth_batch(0, i, args.micro_batch, args.seq_len, vocab)
        e0 = ev()
        loss = wrapper(batch)
        e1 = ev()
        loss.backward()
        e2 = ev()
        engine.step(lr=sched.get_lr(), grad_clip=1.0)
        sched.step()
        e3 = ev()
        rec["REGION_fwd"].append((e0, e1))
        rec["REGION_bwd"].append((e1, e2))
        rec["REGION_opt"].append((e2, e3))

    step(0)  # warmup, untimed
    torch.cuda.synchronize()
    rec.clear()
    for i in range(args.steps):
        step(1 + i)
    torch.cuda.synchronize()

    M = args.micro_batch * args.seq_len
    cfg = wrapper.model.config
    h, ff = cfg.n_embd, getattr(cfg, "n_inner", None) or 4 * cfg.n_embd
    D = h // cfg.n_head
    qkv_n = h + 2 * D * (cfg.num_key_value_heads or 1) if (cfg.num_key_value_heads or 1) != cfg.n_head else 3 * h
    flops = {
        "attn.c_attn": 2 * M * qkv_n * h,
        "attn.c_proj": 2 * M * h * h,
        "mlp.c_fc": 2 * M * ff * h,
        "mlp.c_proj": 2 * M * h * ff,
    }
    for nm, pairs in sorted(rec.items()):
        ms = [a.elapsed_time(b) for a, b in pairs]
        avg = sum(ms) / len(ms)
        line = f"{nm:12s} calls={len(ms):4d} avg={avg:7.3f}ms min={min(ms):7.3f} max={max(ms):7.3f}"
        f = flops.get(nm)
        if f:
            line += f"  -> {f / 1e9 / avg:6.0f} TF"
        print(line)


if __name__ == "__main__":
    main()
