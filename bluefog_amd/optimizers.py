# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""Distributed optimizer wrappers — the training-loop layer.

Reference analog: bluefog/torch/optimizers.py (all five wrapper families,
same factory names, same hook points, same ``synchronize``/
``skip_synchronize``/``num_steps_per_communication`` semantics).

MI355X-native difference: where the reference fires one
nonblocking op per parameter and relies on runtime request fusion
(operations.cc:943-1020), the neighbor/allreduce wrappers here flatten the
model's parameters into a few persistent contiguous *buckets* at
construction (params become views into the bucket storage, so there is no
per-iteration pack/unpack at all) and communicate one bucket per RCCL
group — a handful of ~64 MB xGMI transfers per step, overlapped with
forward+backward, finished by one fused weighted-average HIP kernel each.
"""

import os
import warnings
from contextlib import contextmanager
from enum import Enum
from typing import Dict, List

import torch

from bluefog_amd.utils.env import fusion_threshold_bytes
from bluefog_amd.utils.logging import get_logger

logger = get_logger()



def _attach_timeline_hooks(opt):
    """Model-level FORWARD/BACKWARD host spans (reference analog:
    _register_timeline, optimizers.py:112-163). Together with the GPU comm
    spans emitted by engine.submit, a trace shows whether communication
    overlaps backward."""
    from bluefog_amd.utils.timeline import timeline

    handles = []
    for model in getattr(opt, "_models", []):
        def _pre(mod, inp):
            timeline().start_activity("model.compute", "FORWARD")

        def _post(mod, inp, out):
            timeline().end_activity("model.compute")
            timeline().start_activity("model.compute", "BACKWARD")

        handles.append(model.register_forward_pre_hook(_pre))
        handles.append(model.register_forward_hook(_post))
    return handles


def _detach_timeline_hooks(handles):
    for h in handles:
        try:
            h.remove()
        except Exception:
            pass


def _end_backward_span():
    from bluefog_amd.utils.timeline import timeline

    timeline().end_activity("model.compute")



def _combine_step_torch(kind, b, group, gathered, weights, self_weight):
    """Torch-op replica of the fused combine+step kernels, used when the
    bucket lives on CPU (BLUEFOG_FUSED_STEP=force): lets multi-rank CPU
    soaks drive the EXACT fused control flow the 8-GPU run executes. Math
    mirrors csrc weighted_combine_sgd/adam (verified against torch.optim
    on GPU by tests/test_gpu_fused.py)."""
    flat, grad = b["flat"], b["grad_flat"]
    numel = flat.numel()
    acc = flat.mul(float(self_weight))
    for k, w in enumerate(weights):
        acc.add_(gathered.narrow(0, k * numel, numel), alpha=float(w))
    if kind == "sgd":
        mu = group["momentum"]
        wd = group["weight_decay"]
        lr = group["lr"]
        nesterov = group.get("nesterov", False)
        d = grad.add(acc, alpha=wd) if wd != 0 else grad.clone()
        if mu != 0:
            if b["momentum_flat"] is None:
                b["momentum_flat"] = torch.zeros_like(flat)
            buf = b["momentum_flat"]
            buf.mul_(mu).add_(d)
            d = d.add(buf, alpha=mu) if nesterov else buf
        acc.add_(d, alpha=-lr)
    else:
        b["adam_step"] += 1
        t = b["adam_step"]
        beta1, beta2 = group["betas"]
        eps, lr, wd = group["eps"], group["lr"], group["weight_decay"]
        g = grad.add(acc, alpha=wd) if wd != 0 else grad
        m, v = b["exp_avg"], b["exp_avg_sq"]
        m.mul_(beta1).add_(g, alpha=1 - beta1)
        v.mul_(beta2).addcmul_(g, g, value=1 - beta2)
        bias1 = 1 - beta1 ** t
        bias2 = 1 - beta2 ** t
        denom = (v / bias2).sqrt_().add_(eps)
        acc.addcdiv_(m / bias1, denom, value=-lr)
    flat.copy_(acc)


class CommunicationType(Enum):
    neighbor_allreduce = "neighbor.allreduce"
    hierarchical_neighbor_allreduce = "hierarchical.neighbor.allreduce"
    allreduce = "allreduce"
    empty = "empty"


_warning_message_num_step_per_communication = (
    "Unexpected behavior: num_steps_per_communication is smaller than the "
    "number of forward passes between two optimizer steps. Communication "
    "was triggered more often than configured."
)
_warning_message_backward_pass_per_step = (
    "Unexpected behavior: backward_passes_per_step is smaller than the "
    "number of backward passes between two optimizer steps."
)


def _bf():
    import bluefog_amd as bf

    return bf


def _named_leaf_module(module, parent_name=None):
    """Yield (name, submodule) for every leaf module."""
    if next(module.named_children(), None) is None:
        yield (parent_name, module)
    for name, ch in module.named_children():
        full = name if parent_name is None else f"{parent_name}.{name}"
        yield from _named_leaf_module(ch, full)


def _find_duplicates(lst):
    seen, dup = set(), set()
    for el in lst:
        if el in seen:
            dup.add(el)
        seen.add(el)
    return dup


def _check_named_parameters(optimizer, model):
    """Validate model(s) vs optimizer params; returns (named_parameters,
    models) like the reference (optimizers.py:72-109)."""
    if isinstance(model, torch.nn.Module):
        models = [model]
    elif isinstance(model, (list, tuple)) and all(
        isinstance(m, torch.nn.Module) for m in model
    ):
        models = list(model)
    else:
        raise TypeError("model must be a torch.nn.Module or a list of them")
    named_parameters = []
    for m in models:
        named_parameters.extend(m.named_parameters())
    all_param_ids = {
        id(p) for group in optimizer.param_groups for p in group["params"]
    }
    named_param_ids = {id(p) for _, p in named_parameters}
    unnamed = all_param_ids - named_param_ids
    if unnamed:
        raise ValueError(
            "named_parameters was specified, but one or more model parameters "
            "were not named. To use the distributed optimizer every optimizer "
            "parameter must come from the given model(s)."
        )
    dups = _find_duplicates([k for k, _ in named_parameters])
    if dups:
        raise ValueError(
            f"Parameter names in the model(s) must be unique; duplicates: {sorted(dups)}"
        )
    return named_parameters, models


# ---------------------------------------------------------------------------
# flat parameter buckets
# ---------------------------------------------------------------------------


def _is_dense(t: torch.Tensor) -> bool:
    """True iff the tensor covers its storage span exactly once under some
    permutation of dims (contiguous, channels_last, ...)."""
    if t.numel() == 0:
        return True
    dims = sorted(range(t.dim()), key=lambda d: t.stride(d))
    expect = 1
    for d in dims:
        if t.shape[d] == 1:
            continue
        if t.stride(d) != expect:
            return False
        expect *= t.shape[d]
    return expect == t.numel()


class _FlatBuckets:
    """Flatten parameters into persistent contiguous per-(device,dtype)
    buckets and rebind each ``p.data`` to a view of its bucket."""

    def __init__(self, named_params, bucket_bytes: int):
        self.buckets = []  # each: dict(flat=Tensor, params=[Parameter], name=str)
        groups: Dict[tuple, list] = {}
        seen = set()
        for name, p in named_params:
            if not p.requires_grad or id(p) in seen:
                continue
            seen.add(id(p))
            groups.setdefault((str(p.device), p.dtype), []).append((name, p))
        bidx = 0
        for (_, dtype), plist in groups.items():
            cur: List = []
            cur_bytes = 0
            esize = torch.empty(0, dtype=dtype).element_size()
            for name, p in plist:
                cur.append((name, p))
                cur_bytes += p.numel() * esize
                if cur_bytes >= bucket_bytes:
                    self._seal(cur, bidx)
                    bidx += 1
                    cur, cur_bytes = [], 0
            if cur:
                self._seal(cur, bidx)
                bidx += 1

    def _seal(self, named_plist, bidx: int) -> None:
        total = sum(p.numel() for _, p in named_plist)
        first = named_plist[0][1]
        flat = torch.empty(total, dtype=first.dtype, device=first.device)
        off = 0
        with torch.no_grad():
            for _, p in named_plist:
                n = p.numel()
                seg = flat.narrow(0, off, n)
                if _is_dense(p.data):
                    # preserve the physical layout (e.g. channels_last conv
                    # weights) inside the bucket via a strided view
                    view = torch.as_strided(seg, p.shape, p.data.stride())
                    view.copy_(p.data)
                    p.data = view
                else:
                    seg.copy_(p.data.reshape(-1))
                    p.data = seg.view(p.shape)
                off += n
        self.buckets.append(
            {
                "flat": flat,
                "params": [p for _, p in named_plist],
                "name": f"bucket.{bidx}",
            }
        )

    def __len__(self):
        return len(self.buckets)


# ---------------------------------------------------------------------------
# 1. gradient allreduce (Horovod-style synchronous DP)
# ---------------------------------------------------------------------------


class _DistributedOptimizer(torch.optim.Optimizer):
    """Synchronous gradient averaging over all ranks (reference
    optimizers.py:166-294). The reference fires one nonblocking allreduce
    per parameter and leans on the coordinator's request fusion to batch
    them; here gradients live in persistent flat buckets (views, like the
    AWC fast path) and each bucket's allreduce fires from the backward
    hooks as soon as every gradient in it has accumulated — DDP-style
    bucketing sized for xGMI, still overlapped with the rest of backward."""

    def __init__(self, params, model, backward_passes_per_step=1):
        super(self.__class__, self).__init__(params)
        named_parameters, models = _check_named_parameters(self, model)
        self._models = models
        self._parameter_names = {v: k for k, v in sorted(named_parameters)}
        self._handles = {}  # bucket-name -> handle
        self._synchronized = False
        self._should_synchronize = True
        self._error_encountered = False
        self._backward_passes_per_step = backward_passes_per_step
        self._allreduce_delay = {
            v: self._backward_passes_per_step for _, v in sorted(named_parameters)
        }
        # bucket gradients in reverse registration order ≈ backward firing
        # order, so early buckets complete (and fire) while backward still
        # runs through the front of the model
        self._buckets = _FlatBuckets(
            list(reversed(sorted(named_parameters))), fusion_threshold_bytes()
        )
        self._bucket_of = {}
        self._pending = {}  # bucket name -> set of params still to fire
        with torch.no_grad():
            for i, b in enumerate(self._buckets.buckets):
                b["name"] = f"gradbucket.{i}"
                b["grad_flat"] = torch.zeros_like(b["flat"])
                off = 0
                for q in b["params"]:
                    n = q.numel()
                    seg = b["grad_flat"].narrow(0, off, n)
                    q.grad = torch.as_strided(seg, q.shape, q.data.stride())
                    off += n
                    self._bucket_of[q] = b
        self._timeline_hook_handles = []
        self._use_timeline = False
        if _bf().size() > 1:
            self._register_hooks()

    def _bluefog_base_step(self, closure=None):
        return super(self.__class__, self).step(closure)

    def _register_hooks(self):
        for param_group in self.param_groups:
            for p in param_group["params"]:
                if p.requires_grad:
                    p.register_post_accumulate_grad_hook(self._make_hook())

    def _fire_bucket(self, b):
        self._handles[b["name"]] = _bf().allreduce_nonblocking_(
            b["grad_flat"], average=True, name=b["name"]
        )

    def _make_hook(self):
        def hook(p):
            if self._allreduce_delay[p] <= 0:
                if not self._error_encountered:
                    warnings.warn(_warning_message_backward_pass_per_step)
                    self._error_encountered = True
            self._allreduce_delay[p] -= 1
            if self._allreduce_delay[p] == 0:
                b = self._bucket_of.get(p)
                if b is None:
                    return
                pend = self._pending.setdefault(
                    b["name"], set(id(q) for q in b["params"])
                )
                pend.discard(id(p))
                if not pend and b["name"] not in self._handles:
                    self._fire_bucket(b)

        return hook

    def turn_on_timeline(self):
        self._use_timeline = True
        if not self._timeline_hook_handles:
            self._timeline_hook_handles = _attach_timeline_hooks(self)

    def turn_off_timeline(self):
        self._use_timeline = False
        _detach_timeline_hooks(self._timeline_hook_handles)
        self._timeline_hook_handles = []

    def synchronize(self):
        _end_backward_span()
        bf = _bf()
        with torch.no_grad():
            # only act when a communication round is underway (some bucket
            # fired); with backward_passes_per_step=N the intermediate
            # steps must neither flush nor reset the countdowns (reference
            # optimizers.py:233-248 touches only params with handles)
            # gate on _pending too: a bucket with one never-firing param
            # (unused head) leaves _handles empty even though other params
            # fired — their gradients must still be averaged
            if bf.size() > 1 and (self._handles or self._pending):
                # flush buckets whose parameters never all fired (frozen /
                # unused params): the collective must still run on every rank
                for b in self._buckets.buckets:
                    if b["name"] not in self._handles:
                        self._fire_bucket(b)
                for name, handle in self._handles.items():
                    bf.synchronize(handle)
                for p in self._allreduce_delay:
                    self._allreduce_delay[p] = self._backward_passes_per_step
                self._pending.clear()
        self._handles.clear()
        self._synchronized = True

    @contextmanager
    def skip_synchronize(self):
        self._should_synchronize = False
        try:
            yield
        finally:
            self._should_synchronize = True

    def step(self, closure=None):
        if self._should_synchronize:
            if self._synchronized:
                warnings.warn(
                    "optimizer.step() called without optimizer.skip_synchronize() "
                    "context after optimizer.synchronize(). Consider the "
                    "skip_synchronize() context."
                )
            self.synchronize()
        self._synchronized = False
        return super(self.__class__, self).step(closure)

    def zero_grad(self, set_to_none: bool = True):
        if self._handles:
            raise AssertionError(
                "optimizer.zero_grad() was called after loss.backward() but "
                "before optimizer.step() or optimizer.synchronize()."
            )
        # gradients are views into per-bucket flat buffers
        for b in self._buckets.buckets:
            b["grad_flat"].zero_()
        return None


# ---------------------------------------------------------------------------
# 2. adapt-with-combine (CTA / consensus): the flagship family
# ---------------------------------------------------------------------------


class _DistributedReduceOptimizer(torch.optim.Optimizer):
    """Parameter averaging launched from a model forward hook, overlapped
    with forward+backward; step() = synchronize then base optimizer step
    (reference optimizers.py:297-482). Uses flat buckets (class docstring)."""

    def __init__(self, params, model, communication_type, num_steps_per_communication=1):
        super(self.__class__, self).__init__(params)
        named_parameters, models = _check_named_parameters(self, model)
        # dynamic-topology knobs, read at every hook firing
        self.self_weight = None
        self.src_weights = None
        self.dst_weights = None
        self.src_machine_weights = None
        self.dst_machine_weights = None
        self.enable_topo_check = False

        self._models = models
        self._parameter_names = {v: k for k, v in sorted(named_parameters)}
        self._name_parameters = {k: v for k, v in sorted(named_parameters)}
        self._handles = {}  # bucket-name -> handle
        self._synchronized = False
        self._should_synchronize = True
        self._error_encountered = False
        self._num_steps_per_communication = num_steps_per_communication
        assert isinstance(communication_type, CommunicationType)
        self._communication_type = communication_type
        self._reduce_delay = num_steps_per_communication
        self._timeline_hook_handles = []
        self._use_timeline = False

        self._buckets = _FlatBuckets(named_parameters, fusion_threshold_bytes())
        self._fused = self._init_fused_mode()
        if os.getenv("BLUEFOG_TIMELINE"):
            self.turn_on_timeline()
        if _bf().size() > 1:
            self._register_hooks()

    def _bluefog_base_step(self, closure=None):
        return super(self.__class__, self).step(closure)

    # ------------------------------------------------------------------
    # fused average+step: one CDNA4 kernel per bucket does
    #   p = self_w*p + sum_k w_k*recv_k  followed by the SGD/Adam update
    # in a single pass over HBM (the reference runs torch slice math plus
    # the base optimizer's kernels instead). Gradients are flattened into
    # per-bucket buffers so the kernel sees one contiguous grad stream.
    # ------------------------------------------------------------------
    def _init_fused_mode(self):
        mode = os.environ.get("BLUEFOG_FUSED_STEP", "1")
        if mode in ("0", "false"):
            return None
        if not self._buckets.buckets:
            return None
        dev = self._buckets.buckets[0]["flat"].device
        if dev.type != "cuda" and mode != "force":
            # "force" drives the fused control flow with a torch-op step
            # replica on CPU (multi-rank soak coverage of the GPU path)
            return None
        from bluefog_amd.ops import hip_ext

        if dev.type == "cuda" and not hip_ext.has_extension():
            return None
        if len(self.param_groups) != 1:
            return None
        base = type(self).__mro__[1]
        group = self.param_groups[0]
        if base is torch.optim.SGD and group.get("dampening", 0) == 0:
            kind = "sgd"
        elif base is torch.optim.Adam and not group.get("amsgrad", False):
            kind = "adam"
        else:
            return None
        if self._communication_type not in (
            CommunicationType.neighbor_allreduce,
            CommunicationType.empty,
        ):
            return None
        # flatten gradients: p.grad becomes a view of the bucket's grad
        # buffer with p's physical layout, so autograd accumulates in place
        with torch.no_grad():
            for b in self._buckets.buckets:
                b["grad_flat"] = torch.zeros_like(b["flat"])
                off = 0
                for p in b["params"]:
                    n = p.numel()
                    seg = b["grad_flat"].narrow(0, off, n)
                    p.grad = torch.as_strided(seg, p.shape, p.data.stride())
                    off += n
                if kind == "sgd":
                    b["momentum_flat"] = None  # lazily allocated
                else:
                    b["exp_avg"] = torch.zeros(
                        b["flat"].numel(), dtype=torch.float32, device=dev
                    )
                    b["exp_avg_sq"] = torch.zeros(
                        b["flat"].numel(), dtype=torch.float32, device=dev
                    )
                    b["adam_step"] = 0
        logger.debug("bluefog_amd: fused %s average+step enabled", kind)
        return kind

    def _fused_apply(self, b, works, gathered, weights, self_weight):
        """Make the current stream wait for the bucket's communication, then
        run the fused average+optimizer-step kernel on it."""
        from bluefog_amd import _C

        for w in works:
            w.wait()  # stream-level wait on RCCL work (CPU: host wait)
        group = self.param_groups[0]
        flat, grad = b["flat"], b["grad_flat"]
        gathered = gathered if gathered is not None else flat
        if not flat.is_cuda:
            return _combine_step_torch(self._fused, b, group, gathered,
                                       weights, self_weight)
        if self._fused == "sgd":
            momentum = group["momentum"]
            if momentum != 0 and b["momentum_flat"] is None:
                b["momentum_flat"] = torch.zeros_like(flat)
            _C.weighted_combine_sgd(
                flat,
                float(self_weight),
                gathered,
                [float(w) for w in weights],
                grad,
                b["momentum_flat"] if momentum != 0 else flat.new_empty(0),
                float(group["lr"]),
                float(momentum),
                float(group["weight_decay"]),
                float(group.get("dampening", 0.0)),
                bool(group.get("nesterov", False)),
            )
        else:
            b["adam_step"] += 1
            beta1, beta2 = group["betas"]
            _C.weighted_combine_adam(
                flat,
                float(self_weight),
                gathered,
                [float(w) for w in weights],
                grad,
                b["exp_avg"],
                b["exp_avg_sq"],
                float(group["lr"]),
                float(beta1),
                float(beta2),
                float(group["eps"]),
                float(group["weight_decay"]),
                int(b["adam_step"]),
            )

    def _register_hooks(self):
        # post the parameter exchange from a forward PRE-hook: the sends
        # only READ p.data (forward never mutates parameters), so launching
        # before forward widens the comm/compute overlap window from
        # [backward] to [forward + backward] — on 8 xGMI-linked GPUs the
        # one-peer bucket exchange then hides entirely. The reference fires
        # from post-forward module hooks (optimizers.py:354-392);
        # BLUEFOG_AWC_HOOK=post restores that timing.
        pre = os.environ.get("BLUEFOG_AWC_HOOK", "pre") != "post"
        for model in self._models:
            if pre:
                model.register_forward_pre_hook(self._make_hook())
            else:
                model.register_forward_hook(self._make_hook())

    def _make_hook(self):
        def hook(model, *unused):
            if not model.training:
                return
            if self._handles:
                return  # this step's communication is already in flight
            if self._reduce_delay <= 0:
                if not self._error_encountered:
                    warnings.warn(_warning_message_num_step_per_communication)
                    self._error_encountered = True
            self._reduce_delay -= 1
            if self._reduce_delay == 0:
                self._launch_communication()

        return hook

    def _launch_communication(self):
        bf = _bf()
        if self._fused and self._communication_type == CommunicationType.neighbor_allreduce:
            from bluefog_amd.ops import neighbor

            for b in self._buckets.buckets:
                works, gathered, weights, self_w, keep = (
                    neighbor.post_neighbor_exchange_raw(
                        b["flat"],
                        self.self_weight,
                        self.src_weights,
                        self.dst_weights,
                        self.enable_topo_check,
                    )
                )
                self._handles[b["name"]] = (
                    b,
                    ("fused", works, gathered, weights, self_w, keep),
                )
            return
        for b in self._buckets.buckets:
            name = b["name"]
            if self._communication_type == CommunicationType.allreduce:
                h = bf.allreduce_nonblocking(b["flat"], average=True, name=name)
            elif self._communication_type == CommunicationType.neighbor_allreduce:
                h = bf.neighbor_allreduce_nonblocking(
                    b["flat"],
                    name=name,
                    self_weight=self.self_weight,
                    src_weights=self.src_weights,
                    dst_weights=self.dst_weights,
                    enable_topo_check=self.enable_topo_check,
                )
            elif self._communication_type == CommunicationType.hierarchical_neighbor_allreduce:
                h = bf.hierarchical_neighbor_allreduce_nonblocking(
                    b["flat"],
                    name=name,
                    self_weight=self.self_weight,
                    src_machine_weights=self.src_machine_weights,
                    dst_machine_weights=self.dst_machine_weights,
                    enable_topo_check=self.enable_topo_check,
                )
            elif self._communication_type == CommunicationType.empty:
                h = None
            else:
                raise ValueError("Unsupported CommunicationType encountered.")
            self._handles[name] = (b, h)

    @property
    def communication_type(self):
        return self._communication_type

    @communication_type.setter
    def communication_type(self, value):
        assert isinstance(value, CommunicationType)
        self._communication_type = value

    def turn_on_timeline(self):
        self._use_timeline = True
        if not self._timeline_hook_handles:
            self._timeline_hook_handles = _attach_timeline_hooks(self)

    def turn_off_timeline(self):
        self._use_timeline = False
        _detach_timeline_hooks(self._timeline_hook_handles)
        self._timeline_hook_handles = []

    def synchronize(self):
        _end_backward_span()
        bf = _bf()
        with torch.no_grad():
            for name, (b, handle) in self._handles.items():
                if isinstance(handle, tuple) and handle and handle[0] == "fused":
                    _, works, gathered, weights, self_w, keep = handle
                    self._fused_apply(b, works, gathered, weights, self_w)
                    self._fused_step_done = True
                elif handle is not None:
                    output = bf.synchronize(handle)
                    b["flat"].copy_(output)
            if self._handles:
                # reset the local-step countdown only when communication
                # actually fired this round (reference optimizers.py:437-446
                # resets per-param only for params with handles) — so the
                # "step every iteration, communicate every N" pattern works
                self._reduce_delay = self._num_steps_per_communication
        self._handles.clear()
        self._synchronized = True

    @contextmanager
    def skip_synchronize(self):
        self._should_synchronize = False
        try:
            yield
        finally:
            self._should_synchronize = True

    def step(self, closure=None):
        if self._should_synchronize:
            if self._synchronized:
                warnings.warn(
                    "optimizer.step() called without optimizer.skip_synchronize() "
                    "context after optimizer.synchronize(). Consider the "
                    "skip_synchronize() context."
                )
            self.synchronize()
        self._synchronized = False
        if self._fused is not None:
            # the fused kernel applied avg+update for comm iterations (flag
            # set in synchronize); on communication-free iterations run the
            # same kernel as a plain optimizer step (no neighbors)
            if not getattr(self, "_fused_step_done", False):
                with torch.no_grad():
                    for b in self._buckets.buckets:
                        self._fused_apply(b, [], None, [], 1.0)
            self._fused_step_done = False
            if closure is not None:
                with torch.enable_grad():
                    return closure()
            return None
        return super(self.__class__, self).step(closure)

    def zero_grad(self, set_to_none: bool = True):
        if self._fused is not None:
            # gradients are views into per-bucket flat buffers; zero those
            # (set_to_none would detach autograd from the flat storage)
            for b in self._buckets.buckets:
                b["grad_flat"].zero_()
            return None
        return super(self.__class__, self).zero_grad(set_to_none)


# ---------------------------------------------------------------------------
# 3. adapt-then-combine
# ---------------------------------------------------------------------------


class _DistributedAdaptThenCombineOptimizer(torch.optim.Optimizer):
    """ATC: each parameter's gradient hook first applies the optimizer step
    for that parameter, then launches neighbor averaging of the *updated*
    weights — communication overlaps the rest of backward (reference
    optimizers.py:485-841)."""

    def __init__(self, params, model, communication_type, backward_passes_per_step=1):
        super(self.__class__, self).__init__(params)
        named_parameters, models = _check_named_parameters(self, model)
        self.self_weight = None
        self.src_weights = None
        self.dst_weights = None
        self.src_machine_weights = None
        self.dst_machine_weights = None
        self.enable_topo_check = False

        self._models = models
        self._parameter_names = {v: k for k, v in sorted(named_parameters)}
        self._handles = {}
        self._synchronized = False
        self._should_synchronize = True
        self._error_encountered = False
        self._backward_passes_per_step = backward_passes_per_step
        assert isinstance(communication_type, CommunicationType)
        self._communication_type = communication_type
        self._reduce_delay = {
            v: backward_passes_per_step for _, v in sorted(named_parameters)
        }
        self._step_func = None
        self._timeline_hook_handles = []
        self._use_timeline = False
        # fused-bucket fast path, mirroring the AWC one: when a bucket's
        # gradients have all accumulated, ONE kernel applies the SGD/Adam
        # update to the whole bucket and ONE batched RCCL exchange ships it
        # — O(buckets) instead of O(params) work per step, while keeping
        # ATC's adapt-then-combine ordering and backward overlap
        self._buckets = _FlatBuckets(
            list(reversed(sorted(named_parameters))), fusion_threshold_bytes()
        )
        self._bucket_of = {}
        self._pending = {}
        self._fused = self._init_fused_mode()
        # hooks apply the optimizer update itself (step() never calls the
        # base optimizer), so they must run even at world size 1
        self._register_hooks()

    def _init_fused_mode(self):
        mode = os.environ.get("BLUEFOG_FUSED_STEP", "1")
        if mode in ("0", "false"):
            return None
        if not self._buckets.buckets:
            return None
        dev = self._buckets.buckets[0]["flat"].device
        if dev.type != "cuda" and mode != "force":
            # "force" drives the fused control flow with a torch-op step
            # replica on CPU (multi-rank soak coverage of the GPU path)
            return None
        from bluefog_amd.ops import hip_ext

        if dev.type == "cuda" and not hip_ext.has_extension():
            return None
        if len(self.param_groups) != 1:
            return None
        base = type(self).__mro__[1]
        group = self.param_groups[0]
        if base is torch.optim.SGD and group.get("dampening", 0) == 0:
            kind = "sgd"
        elif base is torch.optim.Adam and not group.get("amsgrad", False):
            kind = "adam"
        else:
            return None
        if self._communication_type not in (
            CommunicationType.neighbor_allreduce,
            CommunicationType.empty,
        ):
            return None
        with torch.no_grad():
            for b in self._buckets.buckets:
                b["grad_flat"] = torch.zeros_like(b["flat"])
                off = 0
                for q in b["params"]:
                    n = q.numel()
                    seg = b["grad_flat"].narrow(0, off, n)
                    q.grad = torch.as_strided(seg, q.shape, q.data.stride())
                    off += n
                    self._bucket_of[q] = b
                if kind == "sgd":
                    b["momentum_flat"] = None
                else:
                    b["exp_avg"] = torch.zeros(
                        b["flat"].numel(), dtype=torch.float32, device=dev
                    )
                    b["exp_avg_sq"] = torch.zeros(
                        b["flat"].numel(), dtype=torch.float32, device=dev
                    )
                    b["adam_step"] = 0
        logger.debug("bluefog_amd: fused ATC %s step enabled", kind)
        return kind

    def _fused_bucket_step(self, b):
        """Plain optimizer step over the bucket (0 neighbors) via the fused
        kernel, then launch the bucket's neighbor exchange."""
        from bluefog_amd import _C

        group = self.param_groups[0]
        flat, grad = b["flat"], b["grad_flat"]
        if not flat.is_cuda:
            _combine_step_torch(self._fused, b, group, flat, [], 1.0)
            return self._fused_bucket_exchange(b)
        if self._fused == "sgd":
            momentum = group["momentum"]
            if momentum != 0 and b["momentum_flat"] is None:
                b["momentum_flat"] = torch.zeros_like(flat)
            _C.weighted_combine_sgd(
                flat, 1.0, flat, [], grad,
                b["momentum_flat"] if momentum != 0 else flat.new_empty(0),
                float(group["lr"]), float(momentum),
                float(group["weight_decay"]),
                float(group.get("dampening", 0.0)),
                bool(group.get("nesterov", False)),
            )
        else:
            b["adam_step"] += 1
            beta1, beta2 = group["betas"]
            _C.weighted_combine_adam(
                flat, 1.0, flat, [], grad, b["exp_avg"], b["exp_avg_sq"],
                float(group["lr"]), float(beta1), float(beta2),
                float(group["eps"]), float(group["weight_decay"]),
                int(b["adam_step"]),
            )
        self._fused_bucket_exchange(b)

    def _fused_bucket_exchange(self, b):
        if (
            _bf().size() > 1
            and self._communication_type == CommunicationType.neighbor_allreduce
        ):
            from bluefog_amd.ops import neighbor

            self._handles[b["name"]] = (
                "fused",
                *neighbor.post_neighbor_exchange_raw(
                    b["flat"],
                    self.self_weight,
                    self.src_weights,
                    self.dst_weights,
                    self.enable_topo_check,
                ),
            )

    def _bluefog_base_step(self, closure=None):
        return super(self.__class__, self).step(closure)

    @property
    def communication_type(self):
        return self._communication_type

    @communication_type.setter
    def communication_type(self, value):
        assert isinstance(value, CommunicationType)
        self._communication_type = value

    def register_step_function(self, step_func):
        """step_func(parameter, gradient, param_group) applies the update for
        one parameter; replaces the built-in sgd/adam/... steps."""
        self._step_func = step_func

    def _register_hooks(self):
        for param_group in self.param_groups:
            for p in param_group["params"]:
                if p.requires_grad:
                    p.register_post_accumulate_grad_hook(
                        self._make_hook(param_group)
                    )

    def _make_hook(self, param_group):
        def hook(p):
            if self._reduce_delay[p] <= 0:
                if not self._error_encountered:
                    warnings.warn(_warning_message_num_step_per_communication)
                    self._error_encountered = True
            self._reduce_delay[p] -= 1
            if self._reduce_delay[p] == 0:
                self._round_fired = True
                with torch.no_grad():
                    if self._fused is not None and self._step_func is None:
                        b = self._bucket_of.get(p)
                        pend = self._pending.setdefault(
                            b["name"], set(id(q) for q in b["params"])
                        )
                        pend.discard(id(p))
                        if not pend and b["name"] not in self._handles:
                            self._fused_bucket_step(b)
                        return
                    if self._step_func is not None:
                        self._step_func(p, p.grad, param_group)
                    else:
                        self._builtin_step(p, p.grad, param_group)
                    self._handles[p] = self._launch_comm_for(p)

        return hook

    # -- parameter-wise optimizer steps (reference optimizers.py:601-760) --
    def _builtin_step(self, p, grad, group):
        if {"momentum", "dampening", "nesterov"}.issubset(group.keys()):
            self._sgd_step(p, grad, group)
        elif {"betas", "eps", "amsgrad"}.issubset(group.keys()):
            self._adam_step(p, grad, group)
        elif {"alpha", "eps", "centered"}.issubset(group.keys()):
            self._rmsprop_step(p, grad, group)
        elif {"lr_decay", "eps"}.issubset(group.keys()):
            self._adagrad_step(p, grad, group)
        elif {"rho", "eps"}.issubset(group.keys()):
            self._adadelta_step(p, grad, group)
        else:
            raise ValueError(
                "Cannot infer the optimizer type for the ATC step; register a "
                "custom step with register_step_function()."
            )

    def _sgd_step(self, p, grad, group):
        wd, momentum, dampening = group["weight_decay"], group["momentum"], group["dampening"]
        nesterov, lr = group["nesterov"], group["lr"]
        d_p = grad
        if wd != 0:
            d_p = d_p.add(p.data, alpha=wd)
        if momentum != 0:
            state = self.state[p]
            buf = state.get("momentum_buffer")
            if buf is None:
                buf = torch.clone(d_p).detach()
                state["momentum_buffer"] = buf
            else:
                buf.mul_(momentum).add_(d_p, alpha=1 - dampening)
            d_p = d_p.add(buf, alpha=momentum) if nesterov else buf
        p.data.add_(d_p, alpha=-lr)

    def _adam_step(self, p, grad, group):
        beta1, beta2 = group["betas"]
        eps, lr, wd = group["eps"], group["lr"], group["weight_decay"]
        amsgrad = group["amsgrad"]
        state = self.state[p]
        if len(state) == 0:
            state["step"] = 0
            state["exp_avg"] = torch.zeros_like(p.data)
            state["exp_avg_sq"] = torch.zeros_like(p.data)
            if amsgrad:
                state["max_exp_avg_sq"] = torch.zeros_like(p.data)
        if wd != 0:
            grad = grad.add(p.data, alpha=wd)
        state["step"] += 1
        t = state["step"]
        exp_avg, exp_avg_sq = state["exp_avg"], state["exp_avg_sq"]
        exp_avg.mul_(beta1).add_(grad, alpha=1 - beta1)
        exp_avg_sq.mul_(beta2).addcmul_(grad, grad, value=1 - beta2)
        bias1 = 1 - beta1 ** t
        bias2 = 1 - beta2 ** t
        if amsgrad:
            torch.maximum(state["max_exp_avg_sq"], exp_avg_sq, out=state["max_exp_avg_sq"])
            denom = (state["max_exp_avg_sq"] / bias2).sqrt_().add_(eps)
        else:
            denom = (exp_avg_sq / bias2).sqrt_().add_(eps)
        p.data.addcdiv_(exp_avg, denom, value=-lr / bias1)

    def _rmsprop_step(self, p, grad, group):
        alpha, eps, lr = group["alpha"], group["eps"], group["lr"]
        wd, momentum, centered = group["weight_decay"], group["momentum"], group["centered"]
        state = self.state[p]
        if len(state) == 0:
            state["step"] = 0
            state["square_avg"] = torch.zeros_like(p.data)
            if momentum > 0:
                state["momentum_buffer"] = torch.zeros_like(p.data)
            if centered:
                state["grad_avg"] = torch.zeros_like(p.data)
        if wd != 0:
            grad = grad.add(p.data, alpha=wd)
        state["step"] += 1
        square_avg = state["square_avg"]
        square_avg.mul_(alpha).addcmul_(grad, grad, value=1 - alpha)
        if centered:
            grad_avg = state["grad_avg"]
            grad_avg.mul_(alpha).add_(grad, alpha=1 - alpha)
            avg = square_avg.addcmul(grad_avg, grad_avg, value=-1).sqrt_().add_(eps)
        else:
            avg = square_avg.sqrt().add_(eps)
        if momentum > 0:
            buf = state["momentum_buffer"]
            buf.mul_(momentum).addcdiv_(grad, avg)
            p.data.add_(buf, alpha=-lr)
        else:
            p.data.addcdiv_(grad, avg, value=-lr)

    def _adagrad_step(self, p, grad, group):
        lr, lr_decay, wd, eps = group["lr"], group["lr_decay"], group["weight_decay"], group["eps"]
        state = self.state[p]
        if len(state) == 0:
            state["step"] = 0
            state["sum"] = torch.zeros_like(p.data)
        if wd != 0:
            grad = grad.add(p.data, alpha=wd)
        state["step"] += 1
        clr = lr / (1 + (state["step"] - 1) * lr_decay)
        state["sum"].addcmul_(grad, grad, value=1)
        std = state["sum"].sqrt().add_(eps)
        p.data.addcdiv_(grad, std, value=-clr)

    def _adadelta_step(self, p, grad, group):
        rho, eps, lr, wd = group["rho"], group["eps"], group["lr"], group["weight_decay"]
        state = self.state[p]
        if len(state) == 0:
            state["step"] = 0
            state["square_avg"] = torch.zeros_like(p.data)
            state["acc_delta"] = torch.zeros_like(p.data)
        if wd != 0:
            grad = grad.add(p.data, alpha=wd)
        state["step"] += 1
        square_avg, acc_delta = state["square_avg"], state["acc_delta"]
        square_avg.mul_(rho).addcmul_(grad, grad, value=1 - rho)
        std = square_avg.add(eps).sqrt_()
        delta = acc_delta.add(eps).sqrt_().div_(std).mul_(grad)
        p.data.add_(delta, alpha=-lr)
        acc_delta.mul_(rho).addcmul_(delta, delta, value=1 - rho)

    # -- communication -----------------------------------------------------
    def _launch_comm_for(self, p):
        bf = _bf()
        if bf.size() == 1:
            return None
        name = self._parameter_names.get(p)
        if self._communication_type == CommunicationType.allreduce:
            return bf.allreduce_nonblocking(p.data, average=True, name=name)
        if self._communication_type == CommunicationType.neighbor_allreduce:
            return bf.neighbor_allreduce_nonblocking(
                p.data,
                name=name,
                self_weight=self.self_weight,
                src_weights=self.src_weights,
                dst_weights=self.dst_weights,
                enable_topo_check=self.enable_topo_check,
            )
        if self._communication_type == CommunicationType.hierarchical_neighbor_allreduce:
            return bf.hierarchical_neighbor_allreduce_nonblocking(
                p.data,
                name=name,
                self_weight=self.self_weight,
                src_machine_weights=self.src_machine_weights,
                dst_machine_weights=self.dst_machine_weights,
                enable_topo_check=self.enable_topo_check,
            )
        if self._communication_type == CommunicationType.empty:
            return None
        raise ValueError("Unsupported CommunicationType encountered.")

    def turn_on_timeline(self):
        self._use_timeline = True
        if not self._timeline_hook_handles:
            self._timeline_hook_handles = _attach_timeline_hooks(self)

    def turn_off_timeline(self):
        self._use_timeline = False
        _detach_timeline_hooks(self._timeline_hook_handles)
        self._timeline_hook_handles = []

    def synchronize(self):
        _end_backward_span()
        bf = _bf()
        from bluefog_amd.ops import hip_ext

        with torch.no_grad():
            if self._fused is not None:
                # flush partially-fired buckets (a frozen/unused param keeps
                # the pending set non-empty so _fused_bucket_step never ran):
                # the fired params must still be stepped and exchanged, and
                # the collective must run on every rank. A fully-fired
                # bucket leaves an EMPTY pending set behind — it already
                # stepped, so only a truthy (non-empty) set may flush.
                for b in self._buckets.buckets:
                    if self._pending.get(b["name"]) and b["name"] not in self._handles:
                        self._fused_bucket_step(b)
            for key, handle in self._handles.items():
                if isinstance(handle, tuple) and handle and handle[0] == "fused":
                    _, works, gathered, weights, self_w, keep = handle
                    for w in works:
                        w.wait()  # stream-ordered
                    flat = next(
                        b["flat"] for b in self._buckets.buckets if b["name"] == key
                    )
                    # in-place fold: flat = self_w*flat + sum w_k*recv_k
                    hip_ext.weighted_combine(flat, flat, self_w, gathered, weights)
                elif handle is not None:
                    output = bf.synchronize(handle)
                    key.set_(output)
            # countdowns restart only after a round in which the hooks
            # applied updates (covers both the fused size-1 case, which has
            # no handles, and the communicate-every-N pattern)
            if getattr(self, "_round_fired", False):
                for p in self._reduce_delay:
                    self._reduce_delay[p] = self._backward_passes_per_step
                self._pending.clear()
            self._round_fired = False
        self._handles.clear()
        self._synchronized = True

    @contextmanager
    def skip_synchronize(self):
        self._should_synchronize = False
        try:
            yield
        finally:
            self._should_synchronize = True

    def step(self, closure=None):
        # the hooks already applied the parameter update; step() only
        # synchronizes the in-flight neighbor averaging
        if self._should_synchronize:
            if self._synchronized:
                warnings.warn(
                    "optimizer.step() called without optimizer.skip_synchronize() "
                    "context after optimizer.synchronize()."
                )
            self.synchronize()
        self._synchronized = False
        if closure is not None:
            with torch.enable_grad():
                return closure()
        return None

    def zero_grad(self, set_to_none: bool = True):
        if self._handles:
            raise AssertionError(
                "optimizer.zero_grad() was called after loss.backward() but "
                "before optimizer.step() or optimizer.synchronize()."
            )
        if self._fused is not None:
            for b in self._buckets.buckets:
                b["grad_flat"].zero_()
            return None
        return super(self.__class__, self).zero_grad(set_to_none)


# ---------------------------------------------------------------------------
# 4. window (async gossip) optimizers
# ---------------------------------------------------------------------------


class _DistributedWinOptimizer(torch.optim.Optimizer):
    """Async gossip through one-sided windows: push (win_put) or pull
    (win_get) neighbor parameters each forward, fold them in with
    win_update at step() (reference optimizers.py:844-1023)."""

    def __init__(self, params, model, num_steps_per_communication, window_prefix, pull_style):
        super(self.__class__, self).__init__(params)
        if pull_style:
            self.src_weights = None
        else:
            self.dst_weights = None
        self.force_barrier = False
        self.window_prefix = window_prefix + "." if window_prefix is not None else ""

        named_parameters, models = _check_named_parameters(self, model)
        self._models = models
        self._pull_style = pull_style
        self._parameter_names = {
            v: self.window_prefix + k for k, v in sorted(named_parameters)
        }
        # MI355X design note: the reference registers ONE WINDOW PER
        # PARAMETER (optimizers.py:933-944) — ~161 windows for ResNet50,
        # each paying per-iteration host work (executor job, kernel launch
        # per destination, store round-trips for versions/mutex). Here the
        # parameters are flattened into a few large flat buckets (the same
        # _FlatBuckets the AWC fast path uses) and each BUCKET gets one
        # window: a handful of big xGMI transfers and O(buckets) host work
        # per iteration instead of O(params).
        self._buckets = _FlatBuckets(named_parameters, fusion_threshold_bytes())
        self._bucket_names = {
            id(b["flat"]): f"{self.window_prefix}winbucket.{i}"
            for i, b in enumerate(self._buckets.buckets)
        }
        self._handles = {}
        self._synchronized = False
        self._should_synchronize = True
        self._error_encountered = False
        self._num_steps_per_communication = num_steps_per_communication
        self._delay = {
            name: num_steps_per_communication for name in self._bucket_names.values()
        }
        self._timeline_hook_handles = []
        self._use_timeline = False
        if _bf().size() > 1:
            self._register_window()
            self._register_hooks()

    def _bluefog_base_step(self, closure=None):
        return super(self.__class__, self).step(closure)

    def __del__(self):
        try:
            self.unregister_window()
        except Exception:
            pass

    def _register_hooks(self):
        for model in self._models:
            hook = self._make_get_hook() if self._pull_style else self._make_put_hook()
            model.register_forward_hook(hook)

    def _fire(self, flat, name):
        bf = _bf()
        if self._pull_style:
            return bf.win_get_nonblocking(
                name=name, src_weights=self.src_weights, require_mutex=True
            )
        return bf.win_put_nonblocking(
            tensor=flat, name=name, dst_weights=self.dst_weights, require_mutex=False
        )

    def _make_hook_impl(self):
        def hook(model, *unused):
            if not model.training:
                return
            for b in self._buckets.buckets:
                name = self._bucket_names[id(b["flat"])]
                if name in self._handles:
                    continue
                if self._delay[name] <= 0:
                    if not self._error_encountered:
                        warnings.warn(_warning_message_num_step_per_communication)
                        self._error_encountered = True
                self._delay[name] -= 1
                if self._delay[name] == 0:
                    self._handles[name] = self._fire(b["flat"], name)

        return hook

    def _make_put_hook(self):
        return self._make_hook_impl()

    def _make_get_hook(self):
        return self._make_hook_impl()

    def _register_window(self):
        bf = _bf()
        for b in self._buckets.buckets:
            name = self._bucket_names[id(b["flat"])]
            if not bf.win_create(b["flat"], name):
                raise ValueError(f"Cannot allocate window for bucket {name}")

    def unregister_window(self):
        bf = _bf()
        if bf.size() <= 1:
            return
        for b in self._buckets.buckets:
            name = self._bucket_names[id(b["flat"])]
            if name in bf.get_current_created_window_names():
                bf.win_free(name)

    def turn_on_timeline(self):
        self._use_timeline = True
        if not self._timeline_hook_handles:
            self._timeline_hook_handles = _attach_timeline_hooks(self)

    def turn_off_timeline(self):
        self._use_timeline = False
        _detach_timeline_hooks(self._timeline_hook_handles)
        self._timeline_hook_handles = []

    @contextmanager
    def skip_synchronize(self):
        self._should_synchronize = False
        try:
            yield
        finally:
            self._should_synchronize = True

    def synchronize(self):
        _end_backward_span()
        bf = _bf()
        with torch.no_grad():
            for name, handle in self._handles.items():
                bf.win_wait(handle)
                self._delay[name] = self._num_steps_per_communication
                # win_update folds the neighbor buffers into the bucket's
                # flat tensor IN PLACE; the parameters are views of it
                bf.win_update(name=name, require_mutex=True)
        self._handles.clear()
        self._synchronized = True

    def step(self, closure=None):
        if self.force_barrier:
            _bf().barrier()
        if self._should_synchronize:
            if self._synchronized:
                warnings.warn(
                    "optimizer.step() called without optimizer.skip_synchronize() "
                    "context after optimizer.synchronize()."
                )
            self.synchronize()
        self._synchronized = False
        return super(self.__class__, self).step(closure)


class _DistributedPushSumOptimizer(torch.optim.Optimizer):
    """Push-sum gossip with weight correction: windows carry the flattened
    parameter plus a trailing scalar p; accumulates with column-stochastic
    weights 1/(outdegree+1); sync divides by the accumulated p (reference
    optimizers.py:1026-1177)."""

    def __init__(self, params, model, num_steps_per_communication):
        super(self.__class__, self).__init__(params)
        bf = _bf()
        outdegree = len(bf.out_neighbor_ranks())
        self.dst_weights = {
            rank: 1.0 / (outdegree + 1) for rank in bf.out_neighbor_ranks()
        }
        self.self_weight = 1.0 / (outdegree + 1)
        self.force_barrier = True

        named_parameters, models = _check_named_parameters(self, model)
        self._models = models
        self._parameter_names = {v: k for k, v in sorted(named_parameters)}
        # bucketed windows, same rationale as _DistributedWinOptimizer: one
        # extended window (bucket + trailing p scalar) per flat bucket
        # instead of the reference's one per parameter
        self._buckets = _FlatBuckets(named_parameters, fusion_threshold_bytes())
        self._bucket_names = {
            id(b["flat"]): f"pushsum.winbucket.{i}"
            for i, b in enumerate(self._buckets.buckets)
        }
        self._handles = {}
        self._named_ps_weights = {}
        self._named_extension_parameters = {}
        self._synchronized = False
        self._should_synchronize = True
        self._error_encountered = False
        self._num_steps_per_communication = num_steps_per_communication
        self._delay = {
            name: num_steps_per_communication for name in self._bucket_names.values()
        }
        self._timeline_hook_handles = []
        self._use_timeline = False
        if bf.size() > 1:
            self._register_window()
            self._register_hooks()

    def _bluefog_base_step(self, closure=None):
        return super(self.__class__, self).step(closure)

    @torch.no_grad()
    def _register_window(self):
        bf = _bf()
        for b in self._buckets.buckets:
            flat = b["flat"]
            name = self._bucket_names[id(flat)]
            ps_weights = torch.Tensor([1.0]).to(flat.dtype).to(flat.device)
            self._named_ps_weights[name] = ps_weights
            extended = torch.cat((flat, ps_weights), 0)
            self._named_extension_parameters[name] = extended
            if not bf.win_create(extended, name, zero_init=True):
                raise ValueError(f"Cannot allocate window for bucket {name}")

    def _register_hooks(self):
        for model in self._models:
            model.register_forward_hook(self._make_hook())

    def _make_hook(self):
        def hook(model, *unused):
            if not model.training:
                return
            bf = _bf()
            for b in self._buckets.buckets:
                name = self._bucket_names[id(b["flat"])]
                if name in self._handles:
                    continue
                if self._delay[name] <= 0:
                    if not self._error_encountered:
                        warnings.warn(_warning_message_num_step_per_communication)
                        self._error_encountered = True
                self._delay[name] -= 1
                if self._delay[name] == 0:
                    with torch.no_grad():
                        ext = self._named_extension_parameters[name]
                        ext[:-1].copy_(b["flat"])
                        ext[-1] = self._named_ps_weights[name].item()
                    self._handles[name] = bf.win_accumulate_nonblocking(
                        tensor=ext,
                        name=name,
                        dst_weights=self.dst_weights,
                        require_mutex=True,
                    )

        return hook

    def turn_on_timeline(self):
        self._use_timeline = True
        if not self._timeline_hook_handles:
            self._timeline_hook_handles = _attach_timeline_hooks(self)

    def turn_off_timeline(self):
        self._use_timeline = False
        _detach_timeline_hooks(self._timeline_hook_handles)
        self._timeline_hook_handles = []

    @contextmanager
    def skip_synchronize(self):
        self._should_synchronize = False
        try:
            yield
        finally:
            self._should_synchronize = True

    def synchronize(self):
        _end_backward_span()
        bf = _bf()
        flats = {self._bucket_names[id(b["flat"])]: b["flat"] for b in self._buckets.buckets}
        with torch.no_grad():
            for name, handle in self._handles.items():
                bf.win_wait(handle)
                self._delay[name] = self._num_steps_per_communication
                ext = self._named_extension_parameters[name]
                # keep self's share, fold in neighbors' accumulations
                ext.mul_(self.self_weight)
                ext = bf.win_update_then_collect(name=name)
                self._named_ps_weights[name].fill_(ext[-1].item())
                # corrected bucket written back in place; params are views
                flats[name].copy_(ext[:-1] / ext[-1])
        self._handles.clear()
        self._synchronized = True

    def step(self, closure=None):
        if self.force_barrier:
            _bf().barrier()
        if self._should_synchronize:
            if self._synchronized:
                warnings.warn(
                    "optimizer.step() called without optimizer.skip_synchronize() "
                    "context after optimizer.synchronize()."
                )
            self.synchronize()
        self._synchronized = False
        return super(self.__class__, self).step(closure)


# ---------------------------------------------------------------------------
# factories (reference optimizers.py:1180-1554)
# ---------------------------------------------------------------------------


def _wrap(optimizer, cls_impl, *args):
    cls = type(optimizer.__class__.__name__, (optimizer.__class__,), dict(cls_impl.__dict__))
    return cls(optimizer.param_groups, *args)


def DistributedGradientAllreduceOptimizer(optimizer, model, backward_passes_per_step=1):
    """Synchronous gradient-averaging DP (Horovod-equivalent)."""
    return _wrap(optimizer, _DistributedOptimizer, model, backward_passes_per_step)


def DistributedAdaptWithCombineOptimizer(
    optimizer,
    model,
    communication_type=CommunicationType.neighbor_allreduce,
    num_steps_per_communication=1,
):
    """CTA/consensus: neighbor-average the parameters (overlapped with
    fwd+bwd), then apply the local gradient."""
    return _wrap(
        optimizer,
        _DistributedReduceOptimizer,
        model,
        communication_type,
        num_steps_per_communication,
    )


def DistributedAdaptThenCombineOptimizer(
    optimizer,
    model,
    communication_type=CommunicationType.neighbor_allreduce,
    backward_passes_per_step=1,
):
    """ATC: per-parameter optimizer step inside the gradient hook, then
    neighbor-average the updated weights."""
    return _wrap(
        optimizer,
        _DistributedAdaptThenCombineOptimizer,
        model,
        communication_type,
        backward_passes_per_step,
    )


def DistributedAllreduceOptimizer(optimizer, model, num_steps_per_communication=1):
    """Deprecated alias: AdaptWithCombine over global allreduce."""
    warnings.warn(
        "DistributedAllreduceOptimizer is deprecated; use "
        "DistributedAdaptWithCombineOptimizer(communication_type="
        "CommunicationType.allreduce)",
        DeprecationWarning,
    )
    return _wrap(
        optimizer,
        _DistributedReduceOptimizer,
        model,
        CommunicationType.allreduce,
        num_steps_per_communication,
    )


def DistributedNeighborAllreduceOptimizer(optimizer, model, num_steps_per_communication=1):
    """Deprecated alias: AdaptWithCombine over neighbor_allreduce."""
    warnings.warn(
        "DistributedNeighborAllreduceOptimizer is deprecated; use "
        "DistributedAdaptWithCombineOptimizer",
        DeprecationWarning,
    )
    return _wrap(
        optimizer,
        _DistributedReduceOptimizer,
        model,
        CommunicationType.neighbor_allreduce,
        num_steps_per_communication,
    )


def DistributedHierarchicalNeighborAllreduceOptimizer(
    optimizer, model, num_steps_per_communication=1
):
    """Deprecated alias: AdaptWithCombine over hierarchical machine-level
    neighbor_allreduce."""
    warnings.warn(
        "DistributedHierarchicalNeighborAllreduceOptimizer is deprecated; use "
        "DistributedAdaptWithCombineOptimizer(communication_type="
        "CommunicationType.hierarchical_neighbor_allreduce)",
        DeprecationWarning,
    )
    return _wrap(
        optimizer,
        _DistributedReduceOptimizer,
        model,
        CommunicationType.hierarchical_neighbor_allreduce,
        num_steps_per_communication,
    )


def DistributedWinPutOptimizer(optimizer, model, num_steps_per_communication=1, window_prefix=None):
    """Async push gossip via one-sided win_put."""
    return _wrap(
        optimizer,
        _DistributedWinOptimizer,
        model,
        num_steps_per_communication,
        window_prefix,
        False,
    )


def DistributedPullGetOptimizer(optimizer, model, num_steps_per_communication=1):
    """Async pull gossip via one-sided win_get."""
    return _wrap(
        optimizer,
        _DistributedWinOptimizer,
        model,
        num_steps_per_communication,
        None,
        True,
    )


def DistributedPushSumOptimizer(optimizer, model, num_steps_per_communication=1):
    """Push-sum with associated-weight correction over win_accumulate."""
    return _wrap(
        optimizer, _DistributedPushSumOptimizer, model, num_steps_per_communication
    )
