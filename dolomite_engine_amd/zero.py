"""ZeRO-2 flat-parameter data-parallel engine — the MI355X-native
reimplementation of the reference's FSDP wrap (distributed/__init__.py:47-236,
SHARD_GRAD_OP semantics) designed for RCCL over xGMI:

  - parameters are repointed into per-bucket flat bf16 (or fp32) buffers,
    packed in reverse registration order (≈ backward completion order);
  - each param's .grad IS a view into the bucket's flat param-dtype grad
    buffer, so autograd accumulates microstep grads in place — no per-param
    copy kernels, exactly like the reference FSDP's FlatParameter grads
    (accumulation in param dtype matches FSDP no_sync semantics);
  - on the sync microstep, each completed bucket's grad is upcast to fp32
    and reduce-scattered (AVG) on a side HIP stream, overlapping the rest
    of backward (fp32 on the wire = the reference's
    communication_dtype=fp32; xGMI: ring RS is per-link bound, buckets
    sized accordingly);
  - each rank owns a contiguous shard of every bucket: fp32 master weights +
    AdamW m/v live only for the shard; the step is ONE fused HIP AdamW kernel
    per bucket (optimization/optimizer.py:74 semantics) writing updated bf16
    params, which are all-gathered back into the bucket on the side stream;
  - grad-norm clipping (train_utils.py:99-103) over shards with a single
    scalar all-reduce.

CPU/gloo test hosts take the same code path with synchronous collectives
(all_reduce + slice instead of reduce_scatter_tensor; gloo has no RS).
"""

import torch
import torch.distributed as dist

from .ops import adamw_step_flat
from .utils import get_rank, get_world_size, is_initialized


class _Bucket:
    def __init__(self, params, dtype, device, world):
        self.params = params  # list[(param, offset)]
        self.numel = sum(p.numel() for p, _ in params)
        self.numel_padded = _padded_numel(self.numel, world)
        self.shard_size = self.numel_padded // world
        self.flat_param = torch.zeros(self.numel_padded, dtype=dtype, device=device)
        # param-dtype flat grads: every p.grad is a view, autograd
        # accumulates in place (FSDP FlatParameter semantics)
        self.flat_grad = torch.zeros(self.numel_padded, dtype=dtype, device=device)
        for p, off in params:
            with torch.no_grad():
                self.flat_param[off : off + p.numel()].copy_(p.data.reshape(-1))
            p.data = self.flat_param[off : off + p.numel()].view(p.shape)
            p.grad = self.flat_grad[off : off + p.numel()].view(p.shape)
        rank = get_rank()
        self.shard_slice = slice(rank * self.shard_size, (rank + 1) * self.shard_size)
        self.master = self.flat_param[self.shard_slice].float()
        self.exp_avg = torch.zeros_like(self.master)
        self.exp_avg_sq = torch.zeros_like(self.master)
        self.shard_grad = torch.zeros_like(self.master)
        self.pending = 0
        self.comm_event = None
        self.wgrad_event = None  # last deferred-wgrad event touching this bucket


def _padded_numel(numel: int, world: int) -> int:
    return ((numel + 4 * world - 1) // (4 * world)) * (4 * world)


class ZeRO2Engine:
    def __init__(
        self,
        model: torch.nn.Module,
        lr: float = 1e-5,
        betas=(0.9, 0.95),
        eps: float = 1e-10,
        weight_decay: float = 0.1,
        bucket_mb: int = 128,
        overlap_comm: bool = True,
        force_collectives: bool = False,
        # Measured perf-NEUTRAL on 1 MI355X (attention kernels at 2 WGs/CU
        # leave no LDS for co-scheduled GEMM workgroups; the streams
        # time-slice): off by default, available for multi-GPU experiments.
        defer_wgrad: bool = False,
    ):
        self.model = model
        self.lr = lr
        self.beta1, self.beta2 = betas
        self.eps = eps
        self.weight_decay = weight_decay
        self.step_count = 0
        self.world = get_world_size()
        self.rank = get_rank()
        self._sync = True
        # force_collectives runs the RCCL reduce-scatter/all-gather +
        # side-stream/event branch even at world size 1 (RS/AG are then
        # identities): lets a single leased GPU execute the exact code the
        # 8-GPU overlap path runs, instead of that branch being dead until
        # a multi-GPU node exists. Covered by tests/test_gpu_model.py.
        self.use_coll = self.world > 1 or (force_collectives and is_initialized())
        self._sq_partials = None   # (1024,) fp32 sqsum scratch (lazy)
        self._sq_out = None        # per-bucket squared-sum outputs (lazy)
        self._clip_coef = None     # device scalar fused into the AdamW grad read

        params = [p for p in model.parameters() if p.requires_grad]
        assert params
        device = params[0].device
        dtype = params[0].dtype
        self.device = device
        self._cast_scratch = None
        self.on_gpu = device.type == "cuda"
        self.overlap = overlap_comm and self.on_gpu and self.use_coll
        self.comm_stream = torch.cuda.Stream() if self.overlap else None
        # wgrad deferral (GPU only): dW GEMMs run on this stream, off the
        # backward critical path; drained before reduce/clip/step
        self.defer_wgrad = defer_wgrad and self.on_gpu
        self.wgrad_stream = torch.cuda.Stream() if self.defer_wgrad else None
        self._wgrad_events = []

        # pack in reverse registration order ≈ backward completion order
        bucket_elems = bucket_mb * 1024 * 1024 // max(dtype.itemsize, 2)
        self.buckets: list[_Bucket] = []
        cur, cur_n = [], 0
        for p in reversed(params):
            cur.append((p, cur_n))
            cur_n += p.numel()
            if cur_n >= bucket_elems:
                self.buckets.append(_Bucket(cur, dtype, device, self.world))
                cur, cur_n = [], 0
        if cur:
            self.buckets.append(_Bucket(cur, dtype, device, self.world))

        self._param_bucket = {}
        for b in self.buckets:
            for p, off in b.params:
                self._param_bucket[p] = (b, off)
        self._hooks = [p.register_post_accumulate_grad_hook(self._grad_hook) for p in params]

        # weight -> (bucket, offset) of its Linear's bias, for wgrad deferral
        self._linear_bias = {}
        for m in model.modules():
            if isinstance(m, torch.nn.Linear) and m.bias is not None:
                if m.weight in self._param_bucket and m.bias in self._param_bucket:
                    self._linear_bias[m.weight] = self._param_bucket[m.bias]
        if self.defer_wgrad:
            _WgradSink.current = self

    # ---- backward-side ----------------------------------------------------

    def set_sync(self, sync: bool) -> None:
        """no_sync semantics (reference train_utils.py:46-76): gradients only
        accumulate on non-boundary microsteps."""
        self._sync = sync
        if sync:
            for b in self.buckets:
                b.pending = len(b.params)
                b.wgrad_event = None

    def _grad_hook(self, p: torch.Tensor) -> None:
        # autograd already accumulated into the bucket view — the hook only
        # tracks bucket completion so the reduce can launch early. If the
        # view was detached externally (zero_grad(set_to_none=True)),
        # autograd created a fresh grad tensor: fold it in and repoint.
        b, off = self._param_bucket[p]
        g = p.grad
        if g is not None:
            view = b.flat_grad[off : off + p.numel()]
            if g.data_ptr() != view.data_ptr():
                view.add_(g.reshape(-1))
                p.grad = view.view(p.shape)
        if self._sync and self.use_coll:
            b.pending -= 1
            if b.pending == 0 and self.overlap:
                self._launch_reduce(b)

    def _comm_cast(self, b: _Bucket) -> torch.Tensor:
        # persistent fp32 staging for the wire copy — stream-ordered reuse
        # across buckets is safe (all reduces run on the one comm stream)
        if b.flat_grad.dtype == torch.float32:
            return b.flat_grad
        if self._cast_scratch is None or self._cast_scratch.numel() < b.numel_padded:
            self._cast_scratch = torch.empty(
                max(bb.numel_padded for bb in self.buckets), dtype=torch.float32, device=self.device
            )
        out = self._cast_scratch[: b.numel_padded]
        out.copy_(b.flat_grad)
        return out

    def _launch_reduce(self, b: _Bucket) -> None:
        self.comm_stream.wait_stream(torch.cuda.current_stream())
        if b.wgrad_event is not None:
            # a deferred wgrad may still be writing this bucket's views on
            # the wgrad stream — the reduce must order after it regardless
            # of which param's completion triggered the launch
            self.comm_stream.wait_event(b.wgrad_event)
        with torch.cuda.stream(self.comm_stream):
            # fp32 on the wire (reference communication_dtype=fp32)
            dist.reduce_scatter_tensor(b.shard_grad, self._comm_cast(b), op=dist.ReduceOp.AVG)
            b.comm_event = torch.cuda.Event()
            b.comm_event.record()

    def _reduce_sync(self, b: _Bucket) -> None:
        if not self.use_coll:
            # single rank, no collectives: the optimizer reads the flat
            # bf16/fp32 grad bucket directly (_grad_src) — the fp32 cast
            # copy this branch used to do was a pure bandwidth pass
            return
        backend = dist.get_backend()
        if backend == "nccl":
            dist.reduce_scatter_tensor(b.shard_grad, self._comm_cast(b), op=dist.ReduceOp.AVG)
        else:  # gloo: no reduce_scatter — all_reduce then slice (same math)
            g32 = b.flat_grad.float()
            dist.all_reduce(g32, op=dist.ReduceOp.SUM)
            b.shard_grad.copy_(g32[b.shard_slice]).div_(self.world)

    # ---- step-side --------------------------------------------------------

    def _finish_reduces(self) -> None:
        self._drain_wgrad()
        for b in self.buckets:
            if self.overlap and b.comm_event is not None:
                torch.cuda.current_stream().wait_event(b.comm_event)
                b.comm_event = None
            else:
                self._reduce_sync(b)

    def _grad_src(self, b: _Bucket) -> torch.Tensor:
        """The tensor the optimizer step reads: the fp32 reduce-scatter
        shard on the collective path, the flat param-dtype grad bucket
        itself otherwise (the AdamW kernel upcasts per element)."""
        return b.shard_grad if self.use_coll else b.flat_grad[b.shard_slice]

    def clip_grad_norm_(self, max_norm: float) -> torch.Tensor:
        """Global grad-norm over shards (train_utils.py:99-103). The clip
        COEFFICIENT is kept as a device scalar and fused into the AdamW
        kernel's grad read (one scalar load instead of a read+write pass
        over every shard grad). Norms use the fixed-order sqsum kernel so
        the clipped step stays bit-deterministic across runs and across
        the plain/collective world-1 paths."""
        if self.on_gpu:
            if self._sq_partials is None:
                self._sq_partials = torch.empty(1024, dtype=torch.float32, device=self.device)
                self._sq_out = torch.empty(len(self.buckets), dtype=torch.float32, device=self.device)
            from .ops import hip  # noqa: PLC0415

            for i, b in enumerate(self.buckets):
                g = self._grad_src(b)
                hip.check(
                    hip.lib().dolomite_sqsum(
                        hip.stream(), hip.ptr(g), g.numel(), hip.ptr(self._sq_partials),
                        hip.ptr(self._sq_out, i), hip.dt(g),
                    ),
                    "sqsum",
                )
            sq = self._sq_out.sum()
        else:
            sq = torch.zeros((), dtype=torch.float32, device=self.device)
            for b in self.buckets:
                sq += self._grad_src(b).float().pow(2).sum()
        if self.world > 1:
            dist.all_reduce(sq, op=dist.ReduceOp.SUM)
        total_norm = sq.sqrt()
        self._clip_coef = None
        if max_norm is not None:
            self._clip_coef = (max_norm / (total_norm + 1e-6)).clamp(max=1.0).to(torch.float32)
        return total_norm

    def step(self, lr: float | None = None, grad_clip: float | None = None) -> torch.Tensor:
        """Reduce (if not already overlapped) -> clip -> fused AdamW on the
        local shard -> all-gather updated params."""
        self._finish_reduces()
        grad_norm = self.clip_grad_norm_(grad_clip) if grad_clip is not None else None
        if grad_clip is None:
            self._clip_coef = None
        self.step_count += 1
        use_lr = self.lr if lr is None else lr
        for b in self.buckets:
            param_out = b.flat_param[b.shard_slice]
            adamw_step_flat(
                b.master, self._grad_src(b), b.exp_avg, b.exp_avg_sq, self.step_count,
                use_lr, self.beta1, self.beta2, self.eps, self.weight_decay,
                param_out=param_out if b.flat_param.dtype != torch.float32 else None,
                grad_scale=self._clip_coef,
            )
            if b.flat_param.dtype == torch.float32:
                param_out.copy_(b.master)
            self._allgather_params(b)
        if self.overlap:
            for b in self.buckets:
                if b.comm_event is not None:
                    torch.cuda.current_stream().wait_event(b.comm_event)
                    b.comm_event = None
        return grad_norm

    def _allgather_params(self, b: _Bucket) -> None:
        if not self.use_coll:
            return
        backend = dist.get_backend()
        if backend == "nccl" and self.overlap:
            self.comm_stream.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(self.comm_stream):
                dist.all_gather_into_tensor(b.flat_param, b.flat_param[b.shard_slice])
                b.comm_event = torch.cuda.Event()
                b.comm_event.record()
        elif backend == "nccl":
            dist.all_gather_into_tensor(b.flat_param, b.flat_param[b.shard_slice])
        else:
            shards = list(b.flat_param.view(self.world, b.shard_size).unbind(0))
            dist.all_gather(shards, b.flat_param[b.shard_slice].clone())
            for i, s in enumerate(shards):
                b.flat_param.view(self.world, b.shard_size)[i].copy_(s)

    def zero_grad(self) -> None:
        self._drain_wgrad()
        for b in self.buckets:
            b.flat_grad.zero_()
            # shard_grad needs no zeroing: the reduce overwrites every slot

    # ---- checkpointing (per-rank shards) ----------------------------------

    def state_dict(self) -> dict:
        return {
            "step": self.step_count,
            "world_size": self.world,
            "buckets": [
                {"master": b.master, "exp_avg": b.exp_avg, "exp_avg_sq": b.exp_avg_sq} for b in self.buckets
            ],
        }

    def load_state_dict(self, sd: dict, all_shards: list | None = None) -> None:
        """Load this rank's optimizer shard. A checkpoint saved at a
        different world size reshards (reference checkpointing.py:105-109
        DCP-reshard equivalence): pass every saved rank's state dict in
        `all_shards` and each bucket's full fp32 state is reassembled,
        re-padded for this world size and re-sliced."""
        if sd["world_size"] != self.world:
            assert all_shards is not None and len(all_shards) == sd["world_size"], (
                f"world size changed {sd['world_size']} -> {self.world}: resharding needs all saved shards"
            )
            self._load_resharded(all_shards)
            return
        assert len(sd["buckets"]) == len(self.buckets)
        self.step_count = sd["step"]
        for b, s in zip(self.buckets, sd["buckets"]):
            b.master.copy_(s["master"])
            b.exp_avg.copy_(s["exp_avg"])
            b.exp_avg_sq.copy_(s["exp_avg_sq"])
            b.flat_param[b.shard_slice].copy_(b.master.to(b.flat_param.dtype))
            self._allgather_params(b)
        self._drain_comm_events()

    def _load_resharded(self, shards: list) -> None:
        saved_world = shards[0]["world_size"]
        assert all(s["world_size"] == saved_world for s in shards)
        assert all(len(s["buckets"]) == len(self.buckets) for s in shards)
        self.step_count = shards[0]["step"]
        for i, b in enumerate(self.buckets):
            for key, dst in (("master", b.master), ("exp_avg", b.exp_avg), ("exp_avg_sq", b.exp_avg_sq)):
                # saved layout: bucket padded for saved_world, split into
                # equal shards; valid data is the first b.numel elements
                full = torch.cat([s["buckets"][i][key].to(dst.device).float() for s in shards])
                flat = torch.zeros(b.numel_padded, dtype=torch.float32, device=dst.device)
                flat[: b.numel].copy_(full[: b.numel])
                dst.copy_(flat[b.shard_slice])
            b.flat_param[b.shard_slice].copy_(b.master.to(b.flat_param.dtype))
            self._allgather_params(b)
        self._drain_comm_events()

    def _drain_comm_events(self) -> None:
        if self.overlap:
            for b in self.buckets:
                if b.comm_event is not None:
                    torch.cuda.current_stream().wait_event(b.comm_event)
                    b.comm_event = None


# ---------------------------------------------------------------------------
# Deferred weight gradients: the wgrad GEMMs (dW = dY^T X) are independent of
# the backward critical path (only dX feeds the next layer's backward), so
# on GPU they run on a dedicated side stream and accumulate STRAIGHT into the
# flat-bucket grad views, overlapped with the rest of backward + attention
# backward kernels (which leave the MFMA pipe ~70% idle). The engine drains
# the stream before any reduce/clip/step touches the grads.
# ---------------------------------------------------------------------------


class _WgradSink:
    current = None  # the live ZeRO2Engine with defer enabled, or None


class DeferredWgradLinear(torch.autograd.Function):
    """F.linear whose backward computes dX inline but hands dW (and db) to
    the engine's wgrad stream. Falls back to inline grads when no engine
    sink is active (CPU, eval, engine-less tests)."""

    @staticmethod
    def forward(ctx, x, weight, bias):
        ctx.save_for_backward(x, weight)
        ctx.has_bias = bias is not None
        return torch.nn.functional.linear(x, weight, bias)

    @staticmethod
    def backward(ctx, dy):
        x, weight = ctx.saved_tensors
        dy2 = dy.reshape(-1, dy.shape[-1])
        x2 = x.reshape(-1, x.shape[-1])
        dx = (dy2 @ weight).view_as(x)
        sink = _WgradSink.current
        if sink is not None and x.is_cuda and sink._defer_wgrad(weight, dy2, x2, ctx.has_bias):
            return dx, None, None
        dw = dy2.t() @ x2
        db = dy2.sum(0) if ctx.has_bias else None
        return dx, dw, db


def _engine_defer_wgrad(self, weight, dy2, x2, has_bias) -> bool:
    if not self.on_gpu or not self.defer_wgrad:
        return False
    ent = self._param_bucket.get(weight)
    if ent is None:
        return False
    bias_ent = self._linear_bias.get(weight) if has_bias else None
    if has_bias and bias_ent is None:
        return False  # unknown bias pairing: let autograd do it inline
    b, off = ent
    s = self.wgrad_stream
    s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        n_out, n_in = dy2.shape[1], x2.shape[1]
        wview = b.flat_grad[off : off + n_out * n_in].view(n_out, n_in)
        # beta=1 accumulate: C += dY^T @ X with fp32 internal accumulation
        wview.addmm_(dy2.t(), x2)
        if bias_ent is not None:
            bb, boff = bias_ent
            bb.flat_grad[boff : boff + n_out].add_(dy2.sum(0))
        ev = torch.cuda.Event()
        ev.record()
    self._wgrad_events.append(ev)
    # the allocator must not recycle these while the side stream reads them
    dy2.record_stream(s)
    x2.record_stream(s)
    for bx in ({b, bias_ent[0]} if bias_ent is not None else {b}):
        bx.wgrad_event = ev
    if self._sync and self.use_coll:
        for bx in ({b, bias_ent[0]} if bias_ent is not None else {b}):
            bx.pending -= 1
            if bx.pending == 0 and self.overlap:
                self._launch_reduce(bx)
    return True


def _engine_drain_wgrad(self) -> None:
    if self._wgrad_events:
        cur = torch.cuda.current_stream()
        for ev in self._wgrad_events:
            cur.wait_event(ev)
        self._wgrad_events.clear()


ZeRO2Engine._defer_wgrad = _engine_defer_wgrad
ZeRO2Engine._drain_wgrad = _engine_drain_wgrad
