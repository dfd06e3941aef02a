"""ORACLE — CPU restatement of the reference hot path. TEST INFRASTRUCTURE ONLY.

This package restates, in plain PyTorch-CPU (fp32) code, the algorithm of the
reference hot path (ibm-granite/dolomite-engine GPTDolomite padding-free
forward/backward + AdamW step). It exists so that the HIP
product path can be parity-tested against an executable specification.

Only the following may import/execute anything under oracle/:
  - tests/
  - __graft_entry__.smoke() (as the checker of the smoke forward)
  - bench.py's cpu_baseline leg (timed as the reported CPU baseline)
The product package (dolomite_engine_amd) must NEVER import this package;
the product path fails loudly when its HIP extension is missing on a GPU box.

Parity pinning: oracle outputs are checked against golden vectors generated
from the reference's own CPU eager path (oracle/gen_golden.py, run in the
build container where /root/reference is mounted; fixtures committed under
tests/golden/). Tolerances mirror the reference's own test suite
(tests/hf_models/single_gpu/hf_models/gpt_dolomite_test.py:59-80 — fp32
logits atol 3e-7, loss atol 1e-5).
"""

from .model import (
    OracleConfig,
    OracleGPTDolomiteForCausalLM,
    adamw_step_ref,
    apply_rope_ref,
    attention_eager_dense_ref,
    attention_varlen_ref,
    cross_entropy_ref,
    layernorm_ref,
    lm_loss_padding_free_ref,
    rmsnorm_ref,
    rope_cos_sin_ref,
    softmax_cross_entropy_fwd_bwd_ref,
)
