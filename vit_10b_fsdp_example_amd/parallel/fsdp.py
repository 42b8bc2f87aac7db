"""From-scratch ZeRO-3 FSDP engine for MI355X.

Capability parity target: torch_xla's XlaFullyShardedDataParallel as the
reference uses it (run_vit_training.py:13,177-181; utils.py:8,29) —
per-module parameter sharding, all-gather before forward and backward,
reshard after forward, reduce-scatter of gradients, full-norm gradient
clipping, shard metadata for offline consolidation, CPU-side wrapping.

The design is MI355X-first rather than a translation:

  * One flat buffer per wrapped unit.  Every parameter of a unit lives
    in ONE flat fp32 master shard (1/world_size each rank, padded), and
    materializes into ONE flat compute-dtype (bf16) buffer per unit.
    Collectives therefore move one ~630 MB message per ViT-10B block —
    the large-message regime where RCCL's multi-link xGMI algorithms
    reach aggregate link bandwidth — instead of 13 per-tensor calls.
  * Mixed precision natively: fp32 master shards + AdamW state, bf16
    gather/compute/reduce-scatter.  compute_dtype=fp32 gives the exact
    reference numerics (used by the CPU parity tests).
  * Overlap by construction: param gathers for the NEXT unit are issued
    asynchronously (prefetch) while the current unit computes, in both
    forward and backward, on a dedicated RCCL communicator; gradient
    reduce-scatters go on a second communicator and are waited only at
    end-of-backward finalization.
  * Storage lifecycle, not graph tricks: the unit's full flat buffer is
    a single allocation whose storage is freed (resize_(0)) on reshard
    and refilled in the pre-backward hook; module params are stable
    leaf *views* into it, so gradient-checkpoint recompute reads the
    refilled weights with zero extra bookkeeping.

Gradient flow: the leaf views accumulate .grad normally; a
post-accumulate-grad hook copies each one into the unit's flat grad
buffer and, when the unit is complete, frees the full params and
launches the async reduce-scatter.  A backward-engine callback
(registered by the root's pre-backward hook) waits for all pending
reductions and installs fp32 mean gradients on the master shards, so
`optimizer.step()` needs no knowledge of any of this.
"""

import os

import torch
import torch.nn as nn

from ..ops import local_sqnorm, scale_
from .comm import CommContext

# Modules currently executing an FSDP forward (root detection).
_EXEC_STACK = []

# How many units ahead to issue async param gathers (both directions).
# 1 = classic one-ahead overlap; raise for 8-GPU runs where a gather
# outlasts a block's compute (see _prefetch_neighbor).
_PREFETCH_DEPTH = max(1, int(os.environ.get("VITFSDP_PREFETCH_DEPTH", "1")))


def _free_storage(t):
    if t.untyped_storage().size() > 0:
        t.untyped_storage().resize_(0)


def _alloc_storage(t, nbytes):
    if t.untyped_storage().size() == 0:
        t.untyped_storage().resize_(nbytes)


class FullyShardedDataParallel(nn.Module):
    WRAPPER_ATTR = "_fsdp_wrapped_module"
    _flat_warned = False

    def __init__(
        self,
        module,
        reshard_after_forward=True,
        flatten_parameters=False,
        compute_dtype=torch.float32,
        device=None,
        prefetch=True,
        shard_on_cpu=False,
    ):
        super().__init__()
        # `flatten_parameters` is accepted for CLI compatibility with the
        # reference (run_vit_training.py:180,359).  This engine is
        # flat-by-design: a unit's parameters always live in one flat
        # buffer, because one large RCCL message per unit is the xGMI-
        # efficient shape.  The flag therefore has no effect either way
        # (documented deviation; consolidation understands our layout) —
        # warn once when --flatten_parameters is requested so the flag is
        # never a silent behavior change.
        if flatten_parameters and not FullyShardedDataParallel._flat_warned:
            FullyShardedDataParallel._flat_warned = True
            import warnings

            warnings.warn(
                "--flatten_parameters accepted for reference-CLI "
                "compatibility: this FSDP engine is flat-by-design (one "
                "flat buffer per unit), so the flag changes nothing",
                stacklevel=2,
            )
        self.reshard_after_forward = reshard_after_forward
        self.flatten_parameters = flatten_parameters
        self.compute_dtype = compute_dtype
        self.prefetch = prefetch
        # Host-offload mode (reference --shard_on_cpu):
        # beyond wrapping the module while it still lives on CPU, the fp32
        # master shard and the AdamW state stay in pinned host memory;
        # each gather stages the shard to the device with an async
        # hipMemcpyAsync from pinned memory, and the reduced grad shard
        # is copied back to the host for the CPU optimizer step.  Device
        # memory then holds only the transient gathered params +
        # activations (the 288 GB sizing lever for 60B-class models).
        self.shard_on_cpu = shard_on_cpu

        self._comm = CommContext.get()
        ws, rank = self._comm.world_size, self._comm.rank

        if device is not None:
            self.device = torch.device(device)
        else:
            p = next(module.parameters(), None)
            self.device = p.device if p is not None else torch.device("cpu")

        if isinstance(module, FullyShardedDataParallel):
            raise ValueError("do not wrap an FSDP instance directly in another FSDP")
        setattr(self, self.WRAPPER_ATTR, module)

        # ---- collect parameters owned by this unit (nested FSDP units
        # manage their own and are pruned from the walk) ----
        infos = []  # [submodule, name, shape, numel, offset, param]
        offset = 0

        def _collect(mod):
            nonlocal offset
            if isinstance(mod, FullyShardedDataParallel):
                return
            for name, p in list(mod.named_parameters(recurse=False)):
                infos.append([mod, name, p.shape, p.numel(), offset, p])
                offset += p.numel()
            for child in mod.children():
                _collect(child)

        _collect(module)
        total = offset
        assert total > 0, "FSDP unit has no parameters to shard"
        self._total_numel = total
        self._padded_numel = -(-total // ws) * ws
        self._shard_numel = self._padded_numel // ws

        # ---- fp32 master shard, built on the module's current device so
        # --shard_on_cpu never materializes full params on the GPU ----
        src_device = infos[0][5].device
        flat = torch.zeros(self._padded_numel, dtype=torch.float32, device=src_device)
        for _, _, _, numel, off, p in infos:
            flat.narrow(0, off, numel).copy_(p.detach().reshape(-1).to(torch.float32))
        if shard_on_cpu:
            shard = flat.narrow(
                0, rank * self._shard_numel, self._shard_numel
            ).clone()
            # VITFSDP_NO_PIN=1 keeps host shards pageable: slower H2D
            # staging, but the safe mode for single-GPU 60B-class runs
            # where pinning ~230 GB of host memory can OOM the box
            # (at ws=8 the per-rank ~29 GB pinned is unproblematic)
            if (self.device.type == "cuda"
                    and os.environ.get("VITFSDP_NO_PIN", "0") != "1"):
                shard = shard.pin_memory()
        else:
            shard = (
                flat.narrow(0, rank * self._shard_numel, self._shard_numel)
                .to(self.device)
                .clone()
            )
        del flat
        self.flat_param = nn.Parameter(shard)

        # bf16 comm mirror of the master shard: FusedAdamW refreshes it
        # in its step epilogue (params carry _bf16_mirror/_mirror_fresh),
        # so the gather path usually skips the fp32->bf16 cast entirely.
        # Device-resident shards only; anything else falls back to the
        # cast in _comm_shard.
        self._mirror = None
        if (compute_dtype == torch.bfloat16 and not shard_on_cpu
                and self.device.type == "cuda"):
            self._mirror = torch.empty(
                self._shard_numel, dtype=torch.bfloat16, device=self.device
            )
            self._mirror.copy_(self.flat_param.data.to(torch.bfloat16))
            self.flat_param._bf16_mirror = self._mirror
            self.flat_param._mirror_fresh = True
        self.register_load_state_dict_post_hook(self._invalidate_mirror_hook)

        # ---- full compute-dtype buffer + stable leaf views ----
        # ws=1 short-circuit: with one rank the "gathered full params"
        # and the bf16 comm mirror are the same numbers, so the full
        # buffer ALIASES the mirror's storage — no gather copy, no
        # separate 2x-params residency, and the optimizer's mirror
        # refresh IS the weight update the next forward reads.  The
        # full buffer is then persistent (never storage-freed).
        self._ws1_alias = ws == 1 and self._mirror is not None
        if self._ws1_alias:
            # alias with its own autograd version counter (mirror
            # refreshes via copy_ must not bump the param views' base)
            self._full_flat = torch.empty(0, dtype=compute_dtype, device=self.device)
            self._full_flat.set_(
                self._mirror.untyped_storage(), 0, (self._padded_numel,)
            )
        else:
            self._full_flat = torch.empty(
                self._padded_numel, dtype=compute_dtype, device=self.device
            )
        # Collective writes go through a separate alias tensor with its
        # own autograd version counter, so refilling the buffer in the
        # pre-backward hook never trips saved-tensor version checks on
        # the views that forward ops saved.
        self._write_alias = torch.empty(0, dtype=compute_dtype, device=self.device)
        self._write_alias.set_(
            self._full_flat.untyped_storage(), 0, (self._padded_numel,)
        )
        self._elem_bytes = self._full_flat.element_size()

        self._param_infos = []  # (submodule, name, shape, numel, offset)
        self._views = []
        with torch.no_grad():
            for mod, name, shape, numel, off, _p in infos:
                view = self._full_flat.narrow(0, off, numel).view(shape).detach()
                view.requires_grad_(True)
                del mod._parameters[name]
                setattr(mod, name, view)
                self._param_infos.append((mod, name, shape, numel, off))
                self._views.append(view)
        if not self._ws1_alias:
            _free_storage(self._full_flat)

        for idx, view in enumerate(self._views):
            view.register_post_accumulate_grad_hook(self._make_grad_hook(idx))

        # flat grad buffer (compute dtype); storage freed between backwards
        self._full_grad = torch.empty(
            self._padded_numel, dtype=compute_dtype, device=self.device
        )
        _free_storage(self._full_grad)

        # runtime state (note: _root_ref may hold an nn.Module and must
        # NOT go through nn.Module.__setattr__, which would register it
        # as a child and create a module cycle)
        self._is_root = None
        object.__setattr__(self, "_root_ref", None)
        self._fwd_order = None  # root only: units in forward execution order
        self._pre_bwd_done = False
        self._grads_arrived = 0
        self._fresh = False  # full params match current master shard
        self._pending_gather = None  # (work, src_ref)
        self._pending_reduce = None  # (work, out_shard)

    # ------------------------------------------------------------------
    # materialization
    # ------------------------------------------------------------------

    @staticmethod
    def _invalidate_mirror_hook(module, incompatible_keys):
        # load_state_dict wrote new master values: the bf16 mirror is stale
        if getattr(module, "_mirror", None) is not None:
            module.flat_param._mirror_fresh = False

    def _comm_shard(self):
        """The shard in comm/compute dtype on the compute device (cast
        from the fp32 master; for --shard_on_cpu this is the pinned-host
        -> device async staging copy).  Uses the AdamW-maintained bf16
        mirror when it is fresh; refreshes it otherwise."""
        if self._mirror is not None:
            if not getattr(self.flat_param, "_mirror_fresh", False):
                self._mirror.copy_(self.flat_param.data.to(torch.bfloat16))
                self.flat_param._mirror_fresh = True
            return self._mirror
        data = self.flat_param.data
        if data.device != self.device:
            data = data.to(self.device, non_blocking=True)
        if data.dtype != self.compute_dtype:
            data = data.to(self.compute_dtype)
        return data

    def _resident(self):
        return self._full_flat.untyped_storage().size() > 0

    def _issue_gather(self, async_op):
        _alloc_storage(self._full_flat, self._padded_numel * self._elem_bytes)
        src = self._comm_shard()
        work = self._comm.all_gather_into(self._write_alias, src, async_op=async_op)
        return (work, src)

    def _materialize(self):
        """Ensure full params are resident and up to date."""
        if self._ws1_alias:
            # full params ARE the mirror; just make sure it is fresh
            # (FusedAdamW refreshes it in its step epilogue, so this is
            # a no-op in the steady state)
            self._comm_shard()
            self._fresh = True
            return
        if self._pending_gather is not None:
            work, _src = self._pending_gather
            self._pending_gather = None
            work.wait()
            self._fresh = True
            return
        if self._resident() and self._fresh:
            return
        work, _src = self._issue_gather(async_op=False)
        work.wait()
        self._fresh = True

    def _prefetch_gather(self):
        """Issue this unit's param gather asynchronously (called by the
        previous unit in execution order while it computes)."""
        if self._ws1_alias:
            return
        if self._pending_gather is not None or (self._resident() and self._fresh):
            return
        self._pending_gather = self._issue_gather(async_op=True)

    def _free_full(self):
        if self._ws1_alias:
            return  # persistent alias of the mirror
        _free_storage(self._full_flat)

    # ------------------------------------------------------------------
    # forward
    # ------------------------------------------------------------------

    def forward(self, *args, **kwargs):
        if self._is_root is None:
            self._is_root = len(_EXEC_STACK) == 0
            if self._is_root:
                self._fwd_order = []
        root = _EXEC_STACK[0] if _EXEC_STACK else self
        object.__setattr__(self, "_root_ref", root)
        if root._fwd_order is not None and self not in root._fwd_order:
            root._fwd_order.append(self)

        _EXEC_STACK.append(self)
        try:
            self._pre_bwd_done = False
            self._grads_arrived = 0
            self._materialize()
            self._prefetch_neighbor(root, direction=+1)
            out = getattr(self, self.WRAPPER_ATTR)(*args, **kwargs)
            grad_flows = self._register_pre_backward(out)
            if self.reshard_after_forward:
                self._free_full()
            elif not grad_flows:
                # no backward will refresh these params; drop them so a
                # later forward re-gathers post-optimizer values
                self._free_full()
                self._fresh = False
        finally:
            _EXEC_STACK.pop()
        return out

    def _prefetch_neighbor(self, root, direction):
        """Issue async gathers for the next _PREFETCH_DEPTH units in
        execution order (+1 = forward, -1 = backward).  Depth 1 hides
        one unit's all-gather behind one unit's compute; on 8 GPUs the
        gather may take longer than a block (xGMI ring latency), so
        VITFSDP_PREFETCH_DEPTH widens the window at the cost of keeping
        that many extra units' full params resident (~600 MB each for
        ViT-10B)."""
        if not self.prefetch or root._fwd_order is None:
            return
        try:
            i = root._fwd_order.index(self)
        except ValueError:
            return
        for step in range(1, _PREFETCH_DEPTH + 1):
            j = i + direction * step
            if 0 <= j < len(root._fwd_order):
                root._fwd_order[j]._prefetch_gather()

    def _register_pre_backward(self, out):
        # hook EVERY grad-carrying output: with a multi-tensor output
        # (e.g. the deferred-residual (hidden, stream) block interface)
        # the engine may route either tensor's gradient into this unit
        # first, and params must be re-gathered before any of them.
        # _pre_bwd_done makes the extra firings no-ops.
        tensors = out if isinstance(out, (tuple, list)) else (out,)
        any_grad = False
        for t in tensors:
            if torch.is_tensor(t) and t.requires_grad:
                t.register_hook(self._pre_backward_hook)
                any_grad = True
        return any_grad

    # ------------------------------------------------------------------
    # backward
    # ------------------------------------------------------------------

    def _pre_backward_hook(self, grad):
        if self._pre_bwd_done:
            return grad
        self._pre_bwd_done = True
        if self._is_root:
            torch.autograd.Variable._execution_engine.queue_callback(
                self._finalize_backward_all
            )
        self._materialize()
        # ws=1 hands _full_grad itself to the optimizer as _comm_grad; if
        # a second backward starts before the step consumed it (gradient
        # accumulation), detach the pending grad before zeroing the buffer
        pending = getattr(self.flat_param, "_comm_grad", None)
        if (
            pending is not None
            and self._full_grad.untyped_storage().size() > 0
            and pending.data_ptr() == self._full_grad.data_ptr()
        ):
            self.flat_param._comm_grad = pending.clone()
        _alloc_storage(self._full_grad, self._padded_numel * self._elem_bytes)
        self._full_grad.zero_()
        # Pre-bind each view's .grad to its slice of the flat grad buffer:
        # AccumulateGrad then adds the backward's gradients straight into
        # the reduce-scatter payload (no separate copy pass).  If autograd
        # ever rebinds .grad out-of-place, the post-accumulate hook
        # detects it by data_ptr and falls back to an explicit copy.
        for (_m, _n, shape, numel, off), view in zip(self._param_infos, self._views):
            view.grad = self._full_grad.narrow(0, off, numel).view(shape)
        # prefetch the unit the backward will need next (previous in
        # forward order)
        if self._root_ref is not None:
            self._prefetch_neighbor(self._root_ref, direction=-1)
        return grad

    def _make_grad_hook(self, idx):
        def hook(view):
            self._on_param_grad(idx, view)

        return hook

    def _on_param_grad(self, idx, view):
        if view.grad is None:
            return
        _mod, _name, _shape, numel, off = self._param_infos[idx]
        _alloc_storage(self._full_grad, self._padded_numel * self._elem_bytes)
        expected_ptr = (
            self._full_grad.data_ptr() + off * self._full_grad.element_size()
        )
        if view.grad.data_ptr() != expected_ptr:
            # autograd rebound .grad out-of-place (or pre-binding was
            # skipped): fall back to an explicit copy into the payload
            self._full_grad.narrow(0, off, numel).copy_(view.grad.reshape(-1))
        view.grad = None
        self._grads_arrived += 1
        if self._grads_arrived == len(self._views):
            self._post_backward()

    def _post_backward(self):
        """All of this unit's param grads are in the flat buffer: free the
        full params and launch the async reduce-scatter."""
        self._free_full()
        self._fresh = False  # master shard will change at optimizer.step
        if self._comm.world_size == 1:
            # single rank: the "reduced shard" IS the flat grad buffer —
            # no copy, no extra allocation (the buffer is handed to the
            # optimizer below and its storage is reused next backward)
            self._pending_reduce = (None, self._full_grad)
            return
        out_shard = torch.empty(
            self._shard_numel, dtype=self.compute_dtype, device=self.device
        )
        work = self._comm.reduce_scatter_into(out_shard, self._full_grad, async_op=True)
        self._pending_reduce = (work, out_shard)

    def _finalize_unit(self):
        """Wait the pending reduction and hand the reduced grad shard to
        the optimizer, releasing buffers.

        The reduced shard stays in comm/compute dtype: it is attached as
        ``flat_param._comm_grad`` with ``flat_param._grad_prescale`` =
        1/world_size, and FusedAdamW folds the prescale (together with
        any deferred clip coefficient) into its fp32 gradient read —
        saving the cast + divide memory passes over every gradient every
        step.  The host-offload path (--shard_on_cpu) keeps the explicit
        fp32 D2H ingestion since the master lives on another device.
        """
        if self._grads_arrived and self._pending_reduce is None:
            # partial grads (frozen subgraph): reduce what we have —
            # missing slices are zeros from the buffer memset.
            self._post_backward()
        if self._pending_reduce is not None:
            work, out_shard = self._pending_reduce
            self._pending_reduce = None
            if work is not None:
                work.wait()
            prescale = 1.0 / self._comm.world_size
            fp32_ingest = (
                out_shard.device != self.flat_param.device  # --shard_on_cpu D2H
                or os.environ.get("VITFSDP_FP32_GRAD_INGEST", "0") == "1"
            )
            if fp32_ingest:
                # explicit fp32 .grad: required for host offload (the
                # master lives on another device) and as the escape hatch
                # for optimizers that only read .grad (anything other
                # than ops.FusedAdamW).  copy=True because at ws=1
                # out_shard aliases _full_grad, which is freed below.
                g = out_shard.to(torch.float32, copy=True)
                if prescale != 1.0:
                    g.mul_(prescale)
                g = g.to(self.flat_param.device)
                if self.flat_param.grad is None:
                    self.flat_param.grad = g
                else:
                    self.flat_param.grad.add_(g)
            elif getattr(self.flat_param, "_comm_grad", None) is not None:
                # gradient accumulation across backwards: fold the new
                # shard into the existing one in fp32 (rare path)
                acc = self.flat_param._comm_grad
                if acc.dtype != torch.float32:
                    acc = acc.to(torch.float32)
                acc.add_(out_shard.to(torch.float32))
                self.flat_param._comm_grad = acc
                self.flat_param._grad_prescale = prescale
            else:
                self.flat_param._comm_grad = out_shard
                self.flat_param._grad_prescale = prescale
            # the optimizer will rewrite the master next; unless it also
            # refreshes the mirror (FusedAdamW does), the mirror is stale
            if self._mirror is not None:
                self.flat_param._mirror_fresh = False
        if (
            getattr(self.flat_param, "_comm_grad", None) is not None
            and self.flat_param._comm_grad.data_ptr() == self._full_grad.data_ptr()
        ):
            pass  # ws=1: the optimizer consumes the buffer in place
        else:
            _free_storage(self._full_grad)
        self._grads_arrived = 0

    def _finalize_backward_all(self):
        for unit in self._all_units():
            unit._finalize_unit()

    def _all_units(self):
        return [m for m in self.modules() if isinstance(m, FullyShardedDataParallel)]

    # ------------------------------------------------------------------
    # grad clipping (reference run_vit_training.py:270 — clip on the FULL
    # gradient norm: local shard sq-norm in one multi-tensor kernel, one
    # scalar all-reduce, one fused scale; shards partition the full
    # gradient so the result equals the unsharded norm)
    # ------------------------------------------------------------------

    def clip_grad_norm_(self, max_norm, norm_type=2.0, defer_scale=False):
        """Clip on the FULL gradient norm.  With defer_scale=True the clip
        coefficient is not applied as a separate pass over the gradients:
        it is left on the master-shard params as a device scalar that
        FusedAdamW folds into its gradient read (one less 2x-total-grad-
        bytes memory pass per step).  Only use defer_scale with an
        optimizer that honors _deferred_grad_scale (ops.FusedAdamW)."""
        assert norm_type == 2.0, "only L2 clipping is supported"
        # gradients arrive either as fp32 .grad (host-offload path) or as
        # comm-dtype _comm_grad with a pending 1/ws prescale that
        # FusedAdamW folds into its read (_finalize_unit)
        units, grads, raw, raw_prescale = [], [], [], 1.0
        for u in self._all_units():
            p = u.flat_param
            if p.grad is not None:
                units.append(u)
                grads.append(p.grad)
            elif getattr(p, "_comm_grad", None) is not None:
                units.append(u)
                raw.append(p._comm_grad)
                raw_prescale = p._grad_prescale
        if not units:
            return torch.zeros((), device=self.device)
        local = None
        if grads:
            local = local_sqnorm(grads)
        if raw:
            # ||prescale * g||^2 == prescale^2 * ||g||^2
            r = local_sqnorm(raw) * (raw_prescale * raw_prescale)
            local = r if local is None else local + r
        if local.device != self.device:
            local = local.to(self.device)
        self._comm.all_reduce_scalar_(local)
        total_norm = local.sqrt()
        # same formula as torch.nn.utils.clip_grad_norm_
        clip_coef = (max_norm / (total_norm + 1e-6)).clamp(max=1.0)
        if defer_scale and self.device.type == "cuda":
            for u in units:
                if getattr(u.flat_param, "_deferred_grad_scale", None) is not None:
                    raise RuntimeError(
                        "clip_grad_norm_(defer_scale=True): the previous "
                        "deferred clip coefficient was never consumed — "
                        "defer_scale requires an optimizer that honors "
                        "_deferred_grad_scale every step (ops.FusedAdamW)"
                    )
                u.flat_param._deferred_grad_scale = clip_coef
        else:
            if grads:
                scale_(grads, clip_coef)
            if raw:
                scale_(raw, clip_coef)
        return total_norm

    # ------------------------------------------------------------------
    # checkpoint metadata (reference utils.py:29)
    # ------------------------------------------------------------------

    @staticmethod
    def _clean_name(name):
        for tag in ("_fsdp_wrapped_module.", "_checkpoint_wrapped_module."):
            name = name.replace(tag, "")
        return name

    def get_shard_metadata(self):
        """Shard-layout record enabling offline consolidation.

        Returns {"world_size", "rank", "shard_info"} where shard_info
        maps each flat-param *state_dict key* (raw, as stored in the
        per-rank checkpoint) to the ordered original-parameter layout
        of that unit's flat buffer.
        """
        shard_info = {}
        for mod_name, mod in self.named_modules():
            if not isinstance(mod, FullyShardedDataParallel):
                continue
            key = (mod_name + "." if mod_name else "") + "flat_param"
            inner = getattr(mod, mod.WRAPPER_ATTR)
            mod_to_prefix = {id(sub): n for n, sub in inner.named_modules()}
            params = []
            for m, name, shape, numel, off in mod._param_infos:
                prefix = mod_to_prefix.get(id(m), "")
                fqn = (prefix + "." if prefix else "") + name
                full_fqn = self._clean_name(
                    (mod_name + "." if mod_name else "") + fqn
                )
                params.append(
                    {
                        "name": full_fqn,
                        "shape": list(shape),
                        "numel": numel,
                        "offset": off,
                    }
                )
            shard_info[key] = {
                "params": params,
                "total_numel": mod._total_numel,
                "padded_numel": mod._padded_numel,
            }
        return {
            "world_size": self._comm.world_size,
            "rank": self._comm.rank,
            "shard_info": shard_info,
        }

    # state_dict: default nn.Module behavior gives exactly the per-rank
    # shard checkpoint (flat_param per unit + buffers); keys carry the
    # wrapper attributes, which consolidation strips via _clean_name.

    def extra_repr(self):
        return (
            f"world_size={self._comm.world_size}, rank={self._comm.rank}, "
            f"total={self._total_numel}, shard={self._shard_numel}, "
            f"compute_dtype={self.compute_dtype}, "
            f"reshard_after_forward={self.reshard_after_forward}"
        )
