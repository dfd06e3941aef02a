"""Golden-vector generator. Runs ONLY in the build container where the
reference is mounted at /root/reference (the GPU box never has it).

Usage:  python -m oracle.gen_golden

Writes small fixtures under tests/golden/ that pin the oracle (and through
it, the HIP product path) to the reference's own CPU eager path:
  - model_{case}.pt : tiny-config state_dict + inputs + logits/loss/grads
    from the reference GPTDolomiteForCausalLM (eager, fp32, dense causal).
  - ops.pt          : per-op goldens from the reference's own modules
    (RMSNorm, LayerNorm path, RoPE tables/apply) and from torch
    (cross_entropy, AdamW — the reference delegates to torch for both:
    model_wrapper/pretraining.py:125, optimization/optimizer.py:74).
  - scheduler.pt    : lr factors from the reference scheduler closures
    (optimization/scheduler.py) at sampled steps.

The committed fixtures + this script are the parity anchor.
"""

import importlib.util
import sys
import types
from pathlib import Path

import torch

GOLDEN_DIR = Path(__file__).resolve().parent.parent / "tests" / "golden"

MODEL_CASES = {
    # name: config kwargs for the reference GPTDolomiteConfig
    "mqa_rope_rmsnorm_gelu": dict(
        attention_head_type="mqa",
        position_embedding_type="rope",
        normalization_function="rmsnorm",
        activation_function="gelu_pytorch_tanh",
        add_bias=True,
    ),
    "gqa_rope_rmsnorm_swiglu": dict(
        attention_head_type="gqa",
        num_key_value_heads=2,
        position_embedding_type="rope",
        normalization_function="rmsnorm",
        activation_function="swiglu",
        add_bias=False,
    ),
    "mha_abs_layernorm_gelu": dict(
        attention_head_type="mha",
        position_embedding_type="learned_absolute",
        normalization_function="layernorm",
        activation_function="gelu_pytorch_tanh",
        add_bias=True,
    ),
    "mqa_rope_rmsnorm_gelu_mup": dict(
        attention_head_type="mqa",
        position_embedding_type="rope",
        normalization_function="rmsnorm",
        activation_function="gelu_pytorch_tanh",
        add_bias=True,
        m_emb=2.0,
        m_width=4.0,
        m_residual=0.5,
        attention_multiplier=0.08,
    ),
}

TINY = dict(
    vocab_size=512,
    n_positions=128,
    n_embd=64,
    n_layer=2,
    n_head=4,
    n_inner=160,
    resid_pdrop=0.0,
    embd_pdrop=0.0,
    attn_pdrop=0.0,
    layer_norm_epsilon=1e-5,
    bos_token_id=0,
    eos_token_id=1,
    pad_token_id=2,
    tie_word_embeddings=False,
)

GRAD_KEYS = [
    "transformer.wte.weight",
    "transformer.h.0.attn.c_attn.weight",
    "transformer.h.0.mlp.c_fc.weight",
    "transformer.h.1.attn.c_proj.weight",
    "transformer.ln_f.weight",
    "lm_head.weight",
]


def gen_model_cases():
    from oracle.ref_shim import make_reference_config, make_reference_model

    torch.manual_seed(1234)
    B, S = 2, 12
    for name, kw in MODEL_CASES.items():
        cfg_kwargs = dict(TINY)
        cfg_kwargs.update(kw)
        cfg = make_reference_config(**cfg_kwargs)
        model = make_reference_model(cfg, "eager")
        model.eval()

        input_ids = torch.randint(0, cfg_kwargs["vocab_size"], (B, S), generator=torch.Generator().manual_seed(42))
        labels = input_ids.clone()

        out = model(input_ids=input_ids, labels=labels)
        out.loss.backward()
        grads = {k: dict(model.named_parameters())[k].grad.detach().clone() for k in GRAD_KEYS if k in dict(model.named_parameters())}

        torch.save(
            dict(
                config=cfg_kwargs,
                state_dict={k: v.detach().clone() for k, v in model.state_dict().items()},
                input_ids=input_ids,
                labels=labels,
                logits=out.logits.detach().clone(),
                loss=out.loss.detach().clone(),
                grads=grads,
            ),
            GOLDEN_DIR / f"model_{name}.pt",
        )
        print(f"model_{name}: loss={out.loss.item():.6f}")


MOE_GRAD_KEYS = [
    "transformer.wte.weight",
    "transformer.h.0.attn.c_attn.weight",
    "transformer.h.0.mlp.gate.weight",
    "transformer.h.0.mlp.c_fc.weight",
    "transformer.h.1.mlp.c_proj.weight",
    "transformer.ln_f.weight",
    "lm_head.weight",
]


def gen_moe_case():
    """Tiny MoE fixture from the reference MoEDolomiteForCausalLM (eager moe,
    eager attention, fp32). Two runs: CLM-only loss, and CLM + mixtral
    load-balancing aux loss with grads (pins the router gradient path)."""
    from oracle.ref_shim import make_reference_moe_config, make_reference_moe_model

    torch.manual_seed(4321)
    cfg_kwargs = dict(TINY)
    cfg_kwargs.update(
        attention_head_type="mqa",
        position_embedding_type="rope",
        normalization_function="rmsnorm",
        activation_function="swiglu",
        add_bias=False,
        num_experts=4,
        num_experts_per_tok=2,
        router_aux_loss_coef=0.01,
    )
    cfg = make_reference_moe_config(**cfg_kwargs)
    model = make_reference_moe_model(cfg, "eager")
    model.eval()

    B, S = 2, 12
    input_ids = torch.randint(0, cfg_kwargs["vocab_size"], (B, S), generator=torch.Generator().manual_seed(99))
    labels = input_ids.clone()

    params = dict(model.named_parameters())

    out_plain = model(input_ids=input_ids, labels=labels, output_router_logits=False)
    out_plain.loss.backward()
    grads = {k: params[k].grad.detach().clone() for k in MOE_GRAD_KEYS if k in params}
    model.zero_grad()

    out = model(input_ids=input_ids, labels=labels, output_router_logits=True)
    out.loss.backward()
    grads_with_aux = {k: params[k].grad.detach().clone() for k in MOE_GRAD_KEYS if k in params}

    torch.save(
        dict(
            config=cfg_kwargs,
            state_dict={k: v.detach().clone() for k, v in model.state_dict().items()},
            input_ids=input_ids,
            labels=labels,
            logits=out.logits.detach().clone(),
            loss=out_plain.loss.detach().clone(),
            loss_with_aux=out.loss.detach().clone(),
            aux_loss=out.aux_loss.detach().clone(),
            grads=grads,
            grads_with_aux=grads_with_aux,
        ),
        GOLDEN_DIR / "model_moe_mqa_rope_rmsnorm_swiglu.pt",
    )
    print(f"model_moe: loss={out_plain.loss.item():.6f} aux={out.aux_loss.item():.6f}")


def gen_ops():
    from oracle.ref_shim import import_reference_hf_models

    import_reference_hf_models()
    from dolomite_engine.hf_models.modeling_utils.normalization.rmsnorm.base import RMSNorm as RefRMSNorm
    from dolomite_engine.hf_models.modeling_utils.position_embedding.rope import RoPE as RefRoPE
    from dolomite_engine.hf_models.modeling_utils.position_embedding.rope import apply_rotary_pos_emb

    g = torch.Generator().manual_seed(7)
    fx = {}

    # RMSNorm fwd+bwd, fp32 and bf16 input
    for dtype in (torch.float32, torch.bfloat16):
        x = torch.randn(33, 96, generator=g).to(dtype).requires_grad_(True)
        m = RefRMSNorm(96, eps=1e-6)
        with torch.no_grad():
            m.weight.copy_(torch.randn(96, generator=g) * 0.1 + 1.0)
        y = m(x)
        dy = torch.randn_like(y)
        y.backward(dy)
        fx[f"rmsnorm_{dtype}"] = dict(
            x=x.detach(), w=m.weight.detach().clone(), eps=1e-6, dy=dy, y=y.detach(), dx=x.grad.detach(), dw=m.weight.grad.detach()
        )

    # LayerNorm fwd+bwd (the reference 'layernorm'/'torch' impl is nn.LayerNorm)
    for dtype in (torch.float32, torch.bfloat16):
        x = torch.randn(33, 96, generator=g).to(dtype).requires_grad_(True)
        m = torch.nn.LayerNorm(96, eps=1e-5).to(dtype)
        with torch.no_grad():
            m.weight.copy_((torch.randn(96, generator=g) * 0.1 + 1.0).to(dtype))
            m.bias.copy_((torch.randn(96, generator=g) * 0.1).to(dtype))
        y = m(x)
        dy = torch.randn_like(y)
        y.backward(dy)
        fx[f"layernorm_{dtype}"] = dict(
            x=x.detach(), w=m.weight.detach().clone(), b=m.bias.detach().clone(), eps=1e-5,
            dy=dy, y=y.detach(), dx=x.grad.detach(), dw=m.weight.grad.detach(), db=m.bias.grad.detach(),
        )

    # RoPE tables + apply, the head dims on the hot path (3B: 80, llama: 128)
    for d in (64, 80, 128):
        rope = RefRoPE(d, max_position_embeddings=96, base=10000)
        cos, sin = rope(96, torch.float32, torch.device("cpu"))
        pos = torch.arange(32)
        q = torch.randn(32, 4, d, generator=g)
        qr = apply_rotary_pos_emb(q, (cos[pos].unsqueeze(1), sin[pos].unsqueeze(1)))
        fx[f"rope_{d}"] = dict(cos=cos, sin=sin, q=q, pos=pos, q_rotated=qr)

    # Cross entropy (torch reference, as the call sites use it)
    logits = torch.randn(64, 512, generator=g)
    labels = torch.randint(0, 512, (64,), generator=g)
    labels[::7] = -100
    logits_leaf = logits.clone().requires_grad_(True)
    loss = torch.nn.functional.cross_entropy(logits_leaf, labels)
    loss.backward()
    fx["cross_entropy"] = dict(logits=logits, labels=labels, loss=loss.detach(), dlogits=logits_leaf.grad.detach())

    # AdamW (torch.optim.AdamW = reference TorchAdamW), 3 steps
    p0 = torch.randn(257, generator=g)
    p = p0.clone().requires_grad_(True)
    opt = torch.optim.AdamW([p], lr=1e-3, betas=(0.9, 0.95), eps=1e-10, weight_decay=0.1)
    grads = [torch.randn(257, generator=g) for _ in range(3)]
    for gr in grads:
        p.grad = gr.clone()
        opt.step()
    st = opt.state[p]
    fx["adamw"] = dict(
        p0=p0, grads=grads, lr=1e-3, beta1=0.9, beta2=0.95, eps=1e-10, wd=0.1,
        p_final=p.detach().clone(), exp_avg=st["exp_avg"].clone(), exp_avg_sq=st["exp_avg_sq"].clone(),
    )

    torch.save(fx, GOLDEN_DIR / "ops.pt")
    print("ops.pt written:", sorted(fx.keys()))


def _load_reference_scheduler_module():
    """Load optimization/scheduler.py + enums.py as an isolated fake package
    (the real top-level package pulls in absent deps like peft)."""
    root = "/root/reference/dolomite_engine"
    pkg = types.ModuleType("ref_iso")
    pkg.__path__ = [root]
    sys.modules["ref_iso"] = pkg

    spec_e = importlib.util.spec_from_file_location("ref_iso.enums", f"{root}/enums.py")
    enums = importlib.util.module_from_spec(spec_e)
    sys.modules["ref_iso.enums"] = enums
    spec_e.loader.exec_module(enums)

    subpkg = types.ModuleType("ref_iso.optimization")
    subpkg.__path__ = [f"{root}/optimization"]
    sys.modules["ref_iso.optimization"] = subpkg

    spec_s = importlib.util.spec_from_file_location("ref_iso.optimization.scheduler", f"{root}/optimization/scheduler.py")
    sched = importlib.util.module_from_spec(spec_s)
    sys.modules["ref_iso.optimization.scheduler"] = sched
    spec_s.loader.exec_module(sched)
    return sched, enums


def gen_scheduler():
    sched_mod, enums = _load_reference_scheduler_module()
    cases = {}
    steps = list(range(0, 101, 5))
    for style in ("constant", "linear", "cosine", "exponential", "power"):
        p = torch.nn.Parameter(torch.zeros(1))
        opt = torch.optim.SGD([p], lr=0.1)
        kwargs = dict(
            num_warmup_steps=10,
            num_constant_steps=0 if style == "power" else 20,
            num_decay_steps=0 if style == "constant" else 60,
            num_training_steps=100,
            lr_decay_style=enums.LRDecaySchedule(style),
            lr_decay_factor=0.1,
            extra_lr_scheduler_args={"a": 4.6, "b": -0.51, "c": 1.0} if style == "power" else {},
        )
        scheduler = sched_mod.get_scheduler(opt, **kwargs)
        lrs = []
        for _ in range(max(steps) + 1):
            lrs.append(opt.param_groups[0]["lr"])
            opt.step()
            scheduler.step()
        cases[style] = dict(kwargs={k: (v.value if hasattr(v, "value") else v) for k, v in kwargs.items()}, base_lr=0.1, lrs_at=steps, lrs=[lrs[s] for s in steps])
    torch.save(cases, GOLDEN_DIR / "scheduler.pt")
    print("scheduler.pt written")


if __name__ == "__main__":
    GOLDEN_DIR.mkdir(parents=True, exist_ok=True)
    import sys as _sys

    if len(_sys.argv) > 1 and _sys.argv[1] == "moe":
        gen_moe_case()
    else:
        gen_model_cases()
        gen_moe_case()
        gen_ops()
        gen_scheduler()
