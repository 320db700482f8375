"""MI355X-native sharded-training runtime (the FSDP/HSDP/DDP engine).

Re-implements the capability of torch-FSDP as used by the reference
(FULL_SHARD / HYBRID_SHARD / NO_SHARD, per-block wrapping, bf16 mixed
precision with fp32 master shards, meta-device init — reference call sites:
main_training_llama.py:82-91, fms_fsdp/policies/*), re-designed for one
8-GPU xGMI clique per node with 288 GB HBM3E per GPU:

- One flat bf16 parameter buffer per transformer block ("unit"); the local
  shard is 1/S of it (S = shard-group size). all_gather_into_tensor over
  RCCL reconstructs the full buffer on a dedicated HIP comm stream with
  configurable lookahead prefetch, overlapped with the previous block's
  compute (SURVEY.md §3.2 hot loop).
- 288 GB HBM lets us default to reshard_after_forward=False for <=13B
  models: parameters stay gathered between forward and backward, so each
  step costs ONE all-gather + ONE reduce-scatter per unit instead of the
  reference's two gathers (torch FSDP re-gathers in backward;
  SURVEY.md §2.3 collectives table).
- Gradients accumulate into a flat bf16 buffer (param.grad views);
  when a unit's backward completes, a bf16 reduce_scatter_tensor runs on a
  second stream and the result is accumulated into an fp32 shard for the
  fused-AdamW HIP kernel. HSDP adds an inter-node all-reduce of the
  reduce-scattered shard.
- Optimizer state (fp32 master + AdamW moments) lives only on the shard.

Collectives go through torch.distributed's "nccl" backend (= RCCL on ROCm)
over the fully-connected 7-link xGMI clique; message sizes are whole-block
shards (≈50 MB for 7B at S=8), large enough to saturate multi-ring RCCL.
On CPU the same code runs on gloo (world_size>1 tested in tests/).
"""

import os
from contextlib import nullcontext
from typing import List, Optional, Type

import torch
import torch.distributed as dist
import torch.nn as nn

from fms_fsdp_amd import ops

_ALIGN = 128  # element alignment for each param inside the flat buffer


def _pad_to(n, k):
    return (n + k - 1) // k * k


class FlatUnit:
    """One sharding unit: a module subtree whose params live in one flat
    bf16 buffer, sharded 1/S per rank."""

    def __init__(self, name: str, module: nn.Module, params: List[nn.Parameter],
                 shard_group, replicate_group, device, param_dtype,
                 reshard_after_forward: bool, param_names=None,
                 reduce_dtype=None):
        self.name = name
        self.module = module
        self.params = params
        self.param_names = param_names or [f"p{i}" for i in range(len(params))]
        self.shard_group = shard_group            # None => no sharding (S=1)
        self.replicate_group = replicate_group    # None => no replication group
        self.device = device
        self.param_dtype = param_dtype
        self.reduce_dtype = reduce_dtype or param_dtype
        self.reshard_after_forward = reshard_after_forward
        self.S = dist.get_world_size(shard_group) if shard_group is not None else 1

        # layout
        self.offsets = []
        off = 0
        for p in params:
            self.offsets.append(off)
            off += _pad_to(p.numel(), _ALIGN)
        self.total = _pad_to(off, _ALIGN * max(self.S, 1))
        self.shard_size = self.total // self.S

        # persistent tensors (allocated in materialize())
        self.flat_param: Optional[torch.Tensor] = None     # (total,) bf16
        self.flat_grad: Optional[torch.Tensor] = None      # (total,) bf16
        self.param_shard: Optional[torch.Tensor] = None    # (shard,) bf16
        self.master_shard: Optional[torch.Tensor] = None   # (shard,) fp32
        self.grad_shard: Optional[torch.Tensor] = None     # (shard,) fp32
        self.exp_avg: Optional[torch.Tensor] = None        # fp32 (adamw m)
        self.exp_avg_sq: Optional[torch.Tensor] = None     # fp32 (adamw v)

        # runtime state
        self._gathered = False
        self._gather_event = None
        self._grad_event = None
        self._grad_countdown = 0
        self._grads_ready_views = False

    # ---------------- construction ----------------

    def materialize(self, src_rank_broadcast: bool):
        """Create shards from the module's current (initialized) params and
        swap param storage to views of the flat buffer."""
        dev = self.device
        with torch.no_grad():
            flat = torch.zeros(self.total, dtype=self.param_dtype, device=dev)
            for p, off in zip(self.params, self.offsets):
                flat[off:off + p.numel()].copy_(p.detach().reshape(-1).to(dev))
            if src_rank_broadcast and dist.is_initialized() and dist.get_world_size() > 1:
                dist.broadcast(flat, src=0)
            rank = dist.get_rank(self.shard_group) if self.shard_group is not None else 0
            if self.S > 1:
                self.param_shard = flat[rank * self.shard_size:(rank + 1) * self.shard_size].clone()
            self.flat_param = flat
            if self.S == 1:
                self.param_shard = self.flat_param  # same storage
            self.master_shard = self.param_shard.float()
            self.exp_avg = torch.zeros_like(self.master_shard)
            self.exp_avg_sq = torch.zeros_like(self.master_shard)
            self.flat_grad = torch.zeros(self.total, dtype=self.param_dtype, device=dev)
            # grads accumulate in param_dtype (autograd writes flat_grad);
            # the reduce-scatter output lives in reduce_dtype (reference
            # MixedPrecision.reduce_dtype, mixed_precision.py:5-27). When
            # S==1 and the dtypes match, the "shard" aliases flat_grad so
            # the optimizer reads autograd's buffer with zero extra
            # memory traffic.
            if self.S > 1 or self.reduce_dtype != self.param_dtype:
                self.grad_shard = torch.zeros(self.shard_size,
                                              dtype=self.reduce_dtype, device=dev)
            else:
                self.grad_shard = self.flat_grad
            self._point_params_into_flat()
            self._set_grad_views()
            self._gathered = True

    def _point_params_into_flat(self):
        with torch.no_grad():
            for p, off in zip(self.params, self.offsets):
                p.data = self.flat_param[off:off + p.numel()].view(p.shape)

    def _set_grad_views(self):
        for p, off in zip(self.params, self.offsets):
            view = self.flat_grad[off:off + p.numel()].view(p.shape)
            if getattr(p, "_direct_wgrad", False):
                # model routes this weight through ops.linear_flat: the
                # wgrad GEMM accumulates straight into the flat buffer
                # (no autograd temp + add pass); p.grad stays None
                p._flat_grad_view = view
                p.grad = None
            else:
                p.grad = view
        self._grads_ready_views = True

    # ---------------- storage control ----------------

    def _free_flat_param(self):
        if self.S == 1:
            return  # flat IS the shard; never free
        self.flat_param.untyped_storage().resize_(0)
        self._gathered = False

    def _alloc_flat_param(self):
        st = self.flat_param.untyped_storage()
        if st.size() == 0:
            st.resize_(self.total * self.flat_param.element_size())

    def free_flat_grad(self, rs_stream):
        """Release the full-size grad buffer after its reduce-scatter has
        been issued (reshard_after_forward memory regime: at 70B the flat
        grads alone are 140 GB if kept resident — they must be transient).
        The buffer is re-allocated and zeroed in prepare_grads() at the
        unit's next backward-pre."""
        if self.S == 1 or self.grad_shard is self.flat_grad:
            return  # S==1: flat_grad may be a pooled view — never resize
        if rs_stream is not None:
            # the caching allocator must not hand this block out until
            # the reduce-scatter reading it on rs_stream has passed
            self.flat_grad.record_stream(rs_stream)
        self.flat_grad.untyped_storage().resize_(0)

    def prepare_grads(self):
        """(Re-)allocate + zero the flat grad buffer if it was freed.
        Zeroes everything (including alignment padding, which
        reduce-scatter reads); direct-wgrad slices are marked fresh so
        their first wgrad GEMM overwrites instead of accumulating."""
        st = self.flat_grad.untyped_storage()
        if st.size() == 0:
            st.resize_(self.total * self.flat_grad.element_size())
            self.flat_grad.zero_()
            for p in self.params:
                if getattr(p, "_direct_wgrad", False):
                    p._wgrad_fresh = True

    # ---------------- collectives ----------------

    def gather(self, comm_stream):
        """Issue async all-gather of the bf16 shard on the comm stream."""
        if self._gathered:
            return
        if self.S == 1:
            self._gathered = True
            return
        if comm_stream is not None:
            # the shard was last written by the optimizer on the default
            # stream: the comm stream must order behind that write
            ev = torch.cuda.Event()
            ev.record(torch.cuda.current_stream())
            comm_stream.wait_event(ev)
        ctx = torch.cuda.stream(comm_stream) if comm_stream is not None else nullcontext()
        with ctx:
            self._alloc_flat_param()
            dist.all_gather_into_tensor(self.flat_param, self.param_shard,
                                        group=self.shard_group)
            if comm_stream is not None:
                self._gather_event = torch.cuda.Event()
                self._gather_event.record(comm_stream)
        self._gathered = True
        self._point_params_into_flat()

    def wait_gather(self):
        if self._gather_event is not None:
            torch.cuda.current_stream().wait_event(self._gather_event)
            self.flat_param.record_stream(torch.cuda.current_stream())
            self._gather_event = None

    def mark_stale(self):
        """After optimizer step the gathered buffers hold old weights."""
        if self.S > 1:
            if self.reshard_after_forward:
                self._free_flat_param()
            else:
                self._gathered = False  # keep storage, re-gather into it

    def reduce_grads(self, rs_stream):
        """reduce-scatter flat grads -> grad_shard in reduce_dtype; HSDP:
        all-reduce the shard across replicas."""
        if rs_stream is not None:
            ev = torch.cuda.Event()
            ev.record(torch.cuda.current_stream())
            rs_stream.wait_event(ev)
        rep = dist.get_world_size(self.replicate_group) if self.replicate_group is not None else 1
        total_dp = self.S * rep  # grads averaged over ALL data-parallel ranks
        ctx = torch.cuda.stream(rs_stream) if rs_stream is not None else nullcontext()
        with ctx:
            if self.S > 1:
                src = self.flat_grad if self.flat_grad.dtype == self.reduce_dtype \
                    else self.flat_grad.to(self.reduce_dtype)
                dist.reduce_scatter_tensor(self.grad_shard, src,
                                           op=dist.ReduceOp.SUM, group=self.shard_group)
            elif self.grad_shard is not self.flat_grad:
                self.grad_shard.copy_(self.flat_grad)   # dtype cast
            if self.replicate_group is not None:
                dist.all_reduce(self.grad_shard, group=self.replicate_group)
            if total_dp > 1:
                self.grad_shard.div_(total_dp)
            if rs_stream is not None:
                self._grad_event = torch.cuda.Event()
                self._grad_event.record(rs_stream)

    def wait_grads(self):
        if self._grad_event is not None:
            torch.cuda.current_stream().wait_event(self._grad_event)
            self._grad_event = None

    # ---------------- optimizer plumbing ----------------

    def publish_master_to_shard(self):
        """master fp32 -> bf16 param shard (source of next all-gather)."""
        with torch.no_grad():
            self.param_shard.copy_(self.master_shard)

    def sharded_numel_unpadded(self):
        return sum(p.numel() for p in self.params)


class ShardedModel(nn.Module):
    """Wraps a model; shards every `block_class` submodule as a unit plus one
    root unit for the remaining params (embedding / final norm / lm_head)."""

    def __init__(self, model: nn.Module, block_class: Type[nn.Module],
                 sharding_strategy: str = "fsdp",
                 device=None,
                 param_dtype=torch.bfloat16,
                 reduce_dtype=torch.bfloat16,
                 reshard_after_forward: bool = False,
                 prefetch_lookahead: int = 1,
                 intra_node_size: Optional[int] = None):
        super().__init__()
        self.model = model
        self.prefetch_lookahead = max(prefetch_lookahead, 0)
        self.reduce_dtype = reduce_dtype
        self._step_trained = False

        world = dist.get_world_size() if dist.is_initialized() else 1
        rank = dist.get_rank() if dist.is_initialized() else 0
        if device is None:
            device = torch.device("cuda", torch.cuda.current_device()) \
                if torch.cuda.is_available() else torch.device("cpu")
        self.device = device
        self.is_cuda = device.type == "cuda"

        # process groups per strategy
        shard_group = replicate_group = None
        if world > 1:
            if sharding_strategy == "fsdp":
                shard_group = dist.group.WORLD
            elif sharding_strategy == "hsdp":
                intra = intra_node_size or int(os.environ.get("LOCAL_WORLD_SIZE", 0)) \
                    or min(world, torch.cuda.device_count() or world)
                if world % intra != 0:
                    intra = world
                mesh_shard, mesh_rep = _build_hsdp_groups(world, rank, intra)
                shard_group, replicate_group = mesh_shard, mesh_rep
                if dist.get_world_size(shard_group) == 1:
                    shard_group = None
                # single node: pure intra-node sharding, no replica dim
                if replicate_group is not None and \
                        dist.get_world_size(replicate_group) == 1:
                    replicate_group = None
            elif sharding_strategy == "ddp":
                shard_group = None
                replicate_group = dist.group.WORLD
            else:
                raise ValueError(f"unknown sharding strategy {sharding_strategy}")
        self.shard_group = shard_group
        self.replicate_group = replicate_group

        # build units: one per block, one root
        blocks = [(n, m) for n, m in model.named_modules() if isinstance(m, block_class)]
        name_of = {id(p): n for n, p in model.named_parameters()}
        block_param_ids = set()
        self.units: List[FlatUnit] = []
        for n, m in blocks:
            params = [p for p in m.parameters() if p.requires_grad]
            block_param_ids.update(id(p) for p in params)
            self.units.append(FlatUnit(n, m, params, shard_group, replicate_group,
                                       device, param_dtype, reshard_after_forward,
                                       param_names=[name_of[id(p)] for p in params],
                                       reduce_dtype=reduce_dtype))
        root_params = [p for p in model.parameters()
                       if p.requires_grad and id(p) not in block_param_ids]
        self.root_unit = None
        if root_params:
            self.root_unit = FlatUnit(
                "_root", model, root_params, shard_group, replicate_group,
                device, param_dtype, False,
                param_names=[name_of[id(p)] for p in root_params],
                reduce_dtype=reduce_dtype)
        self.all_units = ([self.root_unit] if self.root_unit else []) \
            + self.units
        self._unit_of_module = {id(u.module): u for u in self.units}

        if any(p.is_meta for p in model.parameters()):
            # streamed materialization: one unit at a time from meta, so
            # no rank ever holds the whole unsharded model (reference
            # low_cpu_fsdp semantics, param_init.py:9-18; required for
            # 70B: the full fp32 model would be ~280 GB per rank)
            self._materialize_from_meta()
        else:
            # move buffers (e.g. rope tables) to device, keep dtype
            for b in model.buffers():
                b.data = b.data.to(device)
            # materialize shards (broadcast rank-0 init for determinism)
            for u in self.all_units:
                u.materialize(src_rank_broadcast=True)
        self._pool_unit_storage()

        # streams
        self.comm_stream = torch.cuda.Stream() if self.is_cuda else None
        self.rs_stream = torch.cuda.Stream() if self.is_cuda else None

        self._install_hooks()

    def _pool_unit_storage(self):
        """Repack per-unit persistent tensors into single contiguous pools
        (units keep views). The optimizer phase then runs as ONE fused
        AdamW launch + ONE sq-norm launch over the whole model instead of
        2 launches x n_units (the reference pays a foreach launch per
        param group; at 7B/33 units the per-unit scheme measured ~66
        launches per step in profiles/7b_1gpu_step_r01_final.md).

        Pooled in every regime: master_shard, exp_avg, exp_avg_sq (always
        persistent shard-size buffers). S>1 additionally pools
        param_shard / grad_shard (persistent even under
        reshard_after_forward — only the full flat buffers are
        transient). S==1 pools the flat param/grad buffers themselves
        (they are never freed at S==1).
        """
        units = self.all_units
        if not units:
            return
        dev = self.device
        S = units[0].S

        def pool(attr, dtype, sizes):
            total = sum(sizes)
            buf = torch.empty(total, dtype=dtype, device=dev)
            off = 0
            views = []
            for u, n in zip(units, sizes):
                v = buf[off:off + n]
                v.copy_(getattr(u, attr))
                views.append(v)
                off += n
            return buf, views

        sizes = [u.shard_size for u in units]
        for attr, dt in (("master_shard", torch.float32),
                         ("exp_avg", torch.float32),
                         ("exp_avg_sq", torch.float32)):
            buf, views = pool(attr, dt, sizes)
            setattr(self, "_pool_" + attr, buf)
            for u, v in zip(units, views):
                setattr(u, attr, v)
        if S > 1:
            pdt = units[0].param_dtype
            rdt = units[0].reduce_dtype
            self._pool_param_shard, views = pool("param_shard", pdt, sizes)
            for u, v in zip(units, views):
                u.param_shard = v
            self._pool_grad_shard, views = pool("grad_shard", rdt, sizes)
            for u, v in zip(units, views):
                u.grad_shard = v
        else:
            totals = [u.total for u in units]
            pdt = units[0].param_dtype
            self._pool_param_shard, views = pool("flat_param", pdt, totals)
            for u, v in zip(units, views):
                u.flat_param = v
                u.param_shard = v
                u._point_params_into_flat()
            self._pool_grad_shard, gviews = pool("flat_grad", pdt, totals)
            rdt = units[0].reduce_dtype
            sep_reduce = rdt != pdt
            for u, v in zip(units, gviews):
                u.flat_grad = v
                if not sep_reduce:
                    u.grad_shard = v   # keep the S==1 alias invariant
                u._set_grad_views()
            if sep_reduce:
                self._pool_reduce_shard, views = pool("grad_shard", rdt, totals)
                for u, v in zip(units, views):
                    u.grad_shard = v

    def _pooled_grad_norm_src(self):
        """The single tensor clip_grad_norm_ sums squares over."""
        units = self.all_units
        if units and units[0].S == 1 and \
                units[0].grad_shard is not units[0].flat_grad:
            return self._pool_reduce_shard
        return self._pool_grad_shard

    def _materialize_from_meta(self):
        """Materialize units one at a time from a meta-device model.

        Per unit: to_empty() its owning modules on device, run the same
        reset_parameters() the eager path uses (root unit:
        model.reset_root_parameters()), copy into the unit's flat bf16
        buffer (rank-0 broadcast), then swap param storage to flat-buffer
        views so the fp32 originals are freed before the next unit is
        touched. Peak transient memory = ONE unit's init params + its
        flat buffer, never the whole model.
        """
        model, dev = self.model, self.device
        # param id -> (owning module, local name); needed because
        # to_empty() replaces the Parameter objects, so each unit's
        # params list must be refreshed afterwards
        slot_of = {}
        for m in model.modules():
            for n, p in m.named_parameters(recurse=False):
                slot_of[id(p)] = (m, n)
        # buffer-only modules first (rope tables): materialize + rebuild
        for m in model.modules():
            if any(b.is_meta for b in m.buffers(recurse=False)) and \
                    not any(True for _ in m.parameters(recurse=False)):
                m.to_empty(device=dev, recurse=False)
                if hasattr(m, "reset_parameters"):
                    m.reset_parameters()
        for u in self.all_units:   # root first: same RNG order as eager
            slots = [slot_of[id(p)] for p in u.params]
            owners = []
            for m, _ in slots:
                if m not in owners:
                    owners.append(m)
            for m in owners:
                m.to_empty(device=dev, recurse=False)
            if u is self.root_unit and hasattr(model, "reset_root_parameters"):
                model.reset_root_parameters()
            elif u is not self.root_unit and hasattr(u.module, "reset_parameters"):
                u.module.reset_parameters()
            else:
                for m in owners:
                    if hasattr(m, "reset_parameters"):
                        m.reset_parameters()
            # refresh param refs (to_empty swapped the objects), carrying
            # over the direct-wgrad routing flag set in module __init__
            for (m, n), old in zip(slots, u.params):
                if getattr(old, "_direct_wgrad", False):
                    m._parameters[n]._direct_wgrad = True
            u.params = [m._parameters[n] for m, n in slots]
            u.materialize(src_rank_broadcast=True)
        # any remaining non-unit buffers (none have params)
        for b in model.buffers():
            if b.is_meta:
                raise RuntimeError("meta buffer left unmaterialized — give "
                                   "its module a reset_parameters()")
            if b.device != dev:
                b.data = b.data.to(dev)

    # ---------------- hooks / orchestration ----------------

    def _install_hooks(self):
        for i, u in enumerate(self.units):
            u.module.register_forward_pre_hook(self._make_fwd_pre(i))
            u.module.register_forward_hook(self._make_fwd_post(i))
            u.module.register_full_backward_pre_hook(self._make_bwd_pre(i))
        for u in self.all_units:
            for p in u.params:
                if getattr(p, "_direct_wgrad", False):
                    # ops.linear_flat's backward fires this after its
                    # addmm_ into the flat buffer — same countdown as
                    # the post-accumulate hook of ordinary params
                    p._wgrad_done = self._make_wgrad_cb(u)
                else:
                    p.register_post_accumulate_grad_hook(
                        self._make_grad_hook(u))
        # no separate root unit: nothing extra to gather at forward start

    def _make_fwd_pre(self, idx):
        def hook(module, args):
            u = self.units[idx]
            u.gather(self.comm_stream)
            u.wait_gather()
            for j in range(idx + 1, min(idx + 1 + self.prefetch_lookahead,
                                        len(self.units))):
                self.units[j].gather(self.comm_stream)
        return hook

    def _make_fwd_post(self, idx):
        def hook(module, args, out):
            u = self.units[idx]
            if u.reshard_after_forward and torch.is_grad_enabled():
                u._free_flat_param()
            return None
        return hook

    def _make_bwd_pre(self, idx):
        def hook(module, grad_output):
            u = self.units[idx]
            u.prepare_grads()
            if not u._gathered:
                u.gather(self.comm_stream)
            u.wait_gather()
            # prefetch previous blocks (backward order)
            for j in range(idx - 1, max(idx - 1 - self.prefetch_lookahead, -1), -1):
                if self.units[j].reshard_after_forward:
                    self.units[j].gather(self.comm_stream)
            if u._grad_countdown == 0:
                u._grad_countdown = len(u.params)
            return None
        return hook

    def _dec_grad(self, u: FlatUnit):
        if u._grad_countdown == 0:
            u._grad_countdown = len(u.params)
        u._grad_countdown -= 1
        if u._grad_countdown == 0:
            u.reduce_grads(self.rs_stream)
            if u.reshard_after_forward:
                u._free_flat_param()
                u.free_flat_grad(self.rs_stream)

    def _make_grad_hook(self, u: FlatUnit):
        def hook(param):
            self._dec_grad(u)
        return hook

    def _make_wgrad_cb(self, u: FlatUnit):
        def cb():
            self._dec_grad(u)
        return cb

    def forward(self, *args, **kwargs):
        # root params (embedding) needed first; gather root + lookahead
        if self.root_unit is not None:
            self.root_unit.gather(self.comm_stream)
            self.root_unit.wait_gather()
        for j in range(min(self.prefetch_lookahead, len(self.units))):
            self.units[j].gather(self.comm_stream)
        return self.model(*args, **kwargs)

    # ---------------- training utilities ----------------

    def clip_grad_norm_(self, max_norm):
        """Global grad-norm over the grad shards (reference:
        train_utils.py:96 model.clip_grad_norm_). The clip factor is NOT
        applied as an extra pass over the grads — it is folded into the
        fused AdamW kernel (one less 2x-shard-size memory sweep). Thanks
        to the pooled shard storage this is ONE kernel launch."""
        for u in self.all_units:
            u.wait_grads()
        local = ops.sq_norm([self._pooled_grad_norm_src()])
        if self.shard_group is not None:
            dist.all_reduce(local, group=self.shard_group)
        total_norm = local.sqrt()
        self._clip_coef = torch.clamp(max_norm / (total_norm + 1e-6), max=1.0)
        return total_norm

    def zero_grad(self, set_to_none=False):
        self._clip_coef = None
        for u in self.all_units:
            if u.flat_grad.untyped_storage().size() == 0:
                # transient grads (reshard_after_forward): zeroed by
                # prepare_grads() at the unit's backward-pre instead
                continue
            # direct-wgrad slices are never pre-zeroed: their first
            # wgrad of the step overwrites (beta=0 addmm). Only the
            # ordinary-param slices (norm weights, embeddings) need the
            # zero-fill for autograd accumulation. Alignment padding was
            # zeroed at materialize() and is never written afterwards.
            direct = [p for p in u.params
                      if getattr(p, "_direct_wgrad", False)]
            if not direct:
                u.flat_grad.zero_()
            else:
                for p, off in zip(u.params, u.offsets):
                    if getattr(p, "_direct_wgrad", False):
                        p._wgrad_fresh = True
                    else:
                        u.flat_grad[off:off + p.numel()].zero_()
            if u.grad_shard is not u.flat_grad:
                u.grad_shard.zero_()
            if not u._grads_ready_views:
                u._set_grad_views()

    def param_count(self):
        return sum(u.sharded_numel_unpadded() for u in self.all_units)


class ShardedAdamW:
    """Fused AdamW over the fp32 master shards (HIP multi-tensor kernel on
    GPU; reference hyperparams main_training_llama.py:113-115)."""

    def __init__(self, sharded_model: ShardedModel, lr=3e-4, betas=(0.9, 0.95),
                 eps=1e-8, weight_decay=0.1):
        self.m = sharded_model
        self.lr = lr
        self.betas = betas
        self.eps = eps
        self.weight_decay = weight_decay
        self.step_count = 0
        # LambdaLR compatibility
        self.param_groups = [{"lr": lr, "initial_lr": lr}]
        self.defaults = {"lr": lr}

    def step(self):
        self.step_count += 1
        lr = self.param_groups[0]["lr"]
        b1, b2 = self.betas
        gscale = getattr(self.m, "_clip_coef", None)
        m = self.m
        for u in m.all_units:
            u.wait_grads()
        # pooled storage: the whole model updates in ONE fused launch
        pdt = m._pool_param_shard.dtype
        lowp = pdt in (torch.bfloat16, torch.float16)
        published = ops.fused_adamw(
            m._pool_master_shard, m._pooled_grad_norm_src(),
            m._pool_exp_avg, m._pool_exp_avg_sq,
            self.step_count, lr, b1, b2, self.eps, self.weight_decay,
            grad_scale=gscale,
            p_bf16_out=m._pool_param_shard if lowp else None)
        if not published or not lowp:
            with torch.no_grad():
                m._pool_param_shard.copy_(m._pool_master_shard)
        for u in m.all_units:
            u.mark_stale()

    def zero_grad(self, set_to_none=False):
        self.m.zero_grad()

    def state_dict(self):
        return {
            "step": self.step_count,
            "lr": self.param_groups[0]["lr"],
            "units": {u.name: {"exp_avg": u.exp_avg, "exp_avg_sq": u.exp_avg_sq,
                               "master": u.master_shard}
                      for u in self.m.all_units},
        }

    def load_state_dict(self, sd):
        self.step_count = sd["step"]
        self.param_groups[0]["lr"] = sd["lr"]
        for u in self.m.all_units:
            usd = sd["units"][u.name]
            u.exp_avg.copy_(usd["exp_avg"])
            u.exp_avg_sq.copy_(usd["exp_avg_sq"])
            u.master_shard.copy_(usd["master"])
            u.publish_master_to_shard()
            u.mark_stale()


class DynamicGradScaler:
    """Dynamic loss scaling for the fp16 policy (reference fpSixteen,
    mixed_precision.py:5-9; torch.cuda.amp.GradScaler semantics).

    Usage inside the train loop:
        scaler.scale_loss(loss).backward()
        gnorm, stepped = scaler.clip_and_step(model, optimizer, max_norm)

    The unscale (1/scale) is FOLDED into the clip coefficient the fused
    AdamW kernel applies, so no extra pass over the grads is ever made.
    On inf/nan grad norm the step is skipped and the scale backs off;
    after growth_interval consecutive good steps it doubles.
    """

    def __init__(self, enabled=True, init_scale=2.0 ** 16, growth_factor=2.0,
                 backoff_factor=0.5, growth_interval=1000):
        self.enabled = enabled
        self.scale = float(init_scale) if enabled else 1.0
        self.growth_factor = growth_factor
        self.backoff_factor = backoff_factor
        self.growth_interval = growth_interval
        self._good_steps = 0

    def scale_loss(self, loss):
        return loss * self.scale if self.enabled else loss

    def clip_and_step(self, model, optimizer, max_norm):
        """Returns (true grad norm, stepped)."""
        if not self.enabled:
            gnorm = model.clip_grad_norm_(max_norm)
            optimizer.step()
            return gnorm, True
        # clip computed on the SCALED grads against max_norm*scale:
        # coef = clamp(max_norm*scale / norm_s, 1) then /scale gives
        # exactly clamp(max_norm / true_norm, 1/scale) = clip ∘ unscale
        gnorm_scaled = model.clip_grad_norm_(max_norm * self.scale)
        true_norm = gnorm_scaled / self.scale
        if not bool(torch.isfinite(gnorm_scaled)):
            self.scale = max(self.scale * self.backoff_factor, 1.0)
            self._good_steps = 0
            return true_norm, False   # skip the update entirely
        model._clip_coef = model._clip_coef / self.scale
        optimizer.step()
        self._good_steps += 1
        if self._good_steps % self.growth_interval == 0:
            self.scale *= self.growth_factor
        return true_norm, True

    def state_dict(self):
        return {"scale": self.scale, "good_steps": self._good_steps,
                "enabled": self.enabled}

    def load_state_dict(self, sd):
        self.scale = sd["scale"]
        self._good_steps = sd["good_steps"]
        self.enabled = sd["enabled"]


def _build_hsdp_groups(world, rank, intra):
    """Shard intra-node (xGMI clique), replicate inter-node."""
    n_nodes = world // intra
    shard_group = replicate_group = None
    for node in range(n_nodes):
        ranks = list(range(node * intra, (node + 1) * intra))
        g = dist.new_group(ranks)
        if rank in ranks:
            shard_group = g
    for local in range(intra):
        ranks = list(range(local, world, intra))
        g = dist.new_group(ranks)
        if rank in ranks:
            replicate_group = g
    return shard_group, replicate_group
