"""dolomite_engine_amd — MI355X-native data-parallel training engine.

A from-scratch reimplementation of the dolomite-engine hot path
(GPTDolomite padding-free forward/backward + ZeRO-2-sharded AdamW step;
SURVEY.md §8) for AMD Instinct MI355X (gfx950):

  - Python host on PyTorch-ROCm, mirroring the reference's public API
    surface for this path (dolomite_engine.pretrain / finetune entry
    points, YAML arguments, GPTDolomiteForCausalLM).
  - Hand-written HIP/CDNA4 kernels behind a C-ABI shared library
    (dolomite_engine_amd/csrc → libdolomite_hip.so, include/dolomite_hip.h)
    for the hot ops: varlen flash attention, fused RMSNorm(+residual),
    RoPE, fused cross-entropy, fused AdamW.
  - RCCL (torch.distributed backend "nccl" on ROCm) over xGMI for the
    ZeRO-2 flat-parameter sharding collectives.

On a GPU, the HIP extension is REQUIRED for the padding-free path: ops fail
loudly if libdolomite_hip.so is missing — there is no silent eager fallback.
"""

__version__ = "0.1.0"
