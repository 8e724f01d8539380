"""FlatAdamW: flat-buffer fused AdamW for MI355X, ZeRO-shardable.

MI355X-native replacement for the reference optimizer path (torch AdamW +
FusedGradClip ctypes kernels, /root/reference/Src/Main_Scripts/training/
cuda_kernels.py:253-402, fused_grad_clip.cu; DeepSpeed ZeRO stages via
backend_deepspeed.py:129-165). Design:

- every parameter's storage is re-pointed into ONE flat bf16 buffer per
  parameter group (weights), padded to a multiple of 256*world so ZeRO
  shards are equal-size and 16-byte aligned;
- every `p.grad` is a view into ONE flat bf16 grad buffer, so RCCL
  all-reduce / reduce-scatter over xGMI and the grad-norm kernel operate on
  contiguous memory with zero gather/scatter;
- the optimizer step is two HIP kernels per group (l2norm_sq + fused
  clip/AdamW/weight-materialise) with NO host synchronisation — the clip
  scale (and NaN skip) is resolved from device memory inside the kernel;
- with shard_world > 1 (ZeRO-1/2), master/m/v are allocated SHARD-SIZE only
  (memory 6+12/N bytes/param instead of 18) and `step()` updates just this
  rank's shard; the engine (parallel/zero.py) all-gathers the bf16 weights.

In fp32 mode (CPU tests, shard_world == 1) parameters alias the master.
"""

from __future__ import annotations

import math
from typing import Dict, Iterable, List, Optional, Tuple

import torch
import torch.nn as nn

from ..ops import interface as K

NO_DECAY_KEYWORDS = ("bias", "norm", "gate.weight", "mod_router")
ALIGN = 256


def split_decay_groups(model: nn.Module, ep_active: bool = False) -> List[Dict]:
    """Param groups: weight-decayed matrices vs norms/biases/router gates
    (reference backend_fsdp.py:220-252 decay grouping). When expert
    parallelism is active, EP-sharded expert weights form their own group
    (comm='expert'): their grads are NOT replicated across the EP group and
    must not join the global DP all-reduce."""
    from ..parallel.expert_parallel import is_expert_param
    decay, no_decay, expert, tp_shard = [], [], [], []
    for name, p in model.named_parameters():
        if not p.requires_grad:
            continue
        if ep_active and is_expert_param(name):
            # EP-sharded (under TP x EP also TP-sharded): replicas across
            # dp only -> own comm class
            expert.append(p)
        elif getattr(p, "_shard_parallel", False):
            # TP-sharded non-expert weights: replicas across dp x ep
            tp_shard.append(p)
        elif any(k in name for k in NO_DECAY_KEYWORDS) or p.dim() <= 1:
            no_decay.append(p)
        else:
            decay.append(p)
    groups = []
    if decay:
        groups.append({"params": decay, "weight_decay": None})  # default wd
    if no_decay:
        groups.append({"params": no_decay, "weight_decay": 0.0})
    if expert:
        groups.append({"params": expert, "weight_decay": None,
                       "comm": "expert"})
    if tp_shard:
        groups.append({"params": tp_shard, "weight_decay": None,
                       "comm": "tp"})
    return groups


class _FlatGroup:
    """One parameter group flattened into contiguous (padded) buffers."""

    def __init__(self, params: List[torch.Tensor], lr: float, weight_decay: float,
                 shard_rank: int = 0, shard_world: int = 1,
                 comm: str = "dp", offload: bool = False):
        self.params = params
        self.lr = lr
        self.weight_decay = weight_decay
        self.comm = comm          # "dp" (replicated) | "expert" (EP) | "tp" (TP)
        self.shard_rank = shard_rank
        self.shard_world = shard_world
        self.numel = sum(p.numel() for p in params)
        quantum = ALIGN * shard_world
        self.padded = (self.numel + quantum - 1) // quantum * quantum
        self.shard_size = self.padded // shard_world
        self.shard_lo = shard_rank * self.shard_size
        self.shard_hi = self.shard_lo + self.shard_size

        device = params[0].device
        self.dtype = params[0].dtype
        self.device = device
        # bf16 AND fp16 params keep low-precision flat weights + an fp32
        # master copy; fp32 params alias the master directly
        self.low_prec = self.dtype in (torch.bfloat16, torch.float16)
        self.bf16 = self.dtype == torch.bfloat16
        sharded = shard_world > 1

        if self.low_prec:
            self.flat_w = torch.zeros(self.padded, device=device,
                                      dtype=self.dtype)
        else:
            assert not sharded or self.dtype == torch.float32
            self.flat_w = (torch.zeros(self.padded, device=device,
                                       dtype=torch.float32) if sharded else None)
        self.flat_g = torch.zeros(self.padded, device=device, dtype=self.dtype)

        # fill flat weights & re-point params
        off = 0
        self.offsets: List[Tuple[int, int]] = []
        w_master_full = (self.flat_w is None)
        if w_master_full:
            self.master = torch.zeros(self.padded, device=device,
                                      dtype=torch.float32)
        for p in params:
            n = p.numel()
            if w_master_full:
                self.master[off:off + n].copy_(p.data.reshape(-1).float())
                p.data = self.master[off:off + n].view(p.shape)
            else:
                self.flat_w[off:off + n].copy_(p.data.reshape(-1).to(self.dtype))
                p.data = self.flat_w[off:off + n].view(p.shape)
            p.grad = self.flat_g[off:off + n].view(p.shape)
            self.offsets.append((off, n))
            off += n

        # optimizer state: shard-size (== padded when world==1).
        # offload=True keeps master/m/v in host (pinned) memory and steps on
        # the CPU — the MI355X-native stand-in for the reference's DeepSpeed
        # optimizer offload (backend_deepspeed.py:129-165) and ColossalAI
        # cpu_adam (extensions/csrc/cuda/cpu_adam.cpp). Unavailable when
        # params alias the master (fp32, unsharded): the model itself would
        # leave the GPU.
        self.offload = bool(offload) and not w_master_full
        pin = device.type == "cuda"
        state_dev = torch.device("cpu") if self.offload else device
        if not w_master_full:
            shard = self.flat_w[self.shard_lo:self.shard_hi].float()
            self.master = (shard.cpu().pin_memory() if self.offload and pin
                           else shard.cpu() if self.offload else shard)
        state_size = self.padded if w_master_full else self.shard_size
        self.m = torch.zeros(state_size, device=state_dev, dtype=torch.float32)
        self.v = torch.zeros_like(self.m)
        self._master_is_params = w_master_full
        if self.offload and pin:
            # pinned staging for async D2H grad / H2D weight transfers
            self._g_stage = torch.zeros(self.shard_size, dtype=self.dtype,
                                        pin_memory=True)
            self._w_stage = torch.zeros(self.shard_size, dtype=self.dtype,
                                        pin_memory=True)

    # ---- views used by step() -------------------------------------------
    def update_grad(self) -> torch.Tensor:
        """The grad slice this rank's update consumes."""
        if self._master_is_params:
            return self.flat_g
        return self.flat_g[self.shard_lo:self.shard_hi]

    def update_weight_out(self) -> Optional[torch.Tensor]:
        if self._master_is_params:
            return None  # params alias master
        return self.flat_w[self.shard_lo:self.shard_hi]

    def weight_view(self) -> torch.Tensor:
        return self.master if self._master_is_params else self.flat_w

    def grad_view(self) -> torch.Tensor:
        return self.flat_g


class FlatAdamW:
    """torch-optimizer-like interface over flat fused AdamW (ZeRO-aware)."""

    def __init__(self, model_or_groups, lr: float = 1e-4,
                 betas: Tuple[float, float] = (0.9, 0.95), eps: float = 1e-8,
                 weight_decay: float = 0.01, max_grad_norm: float = 1.0,
                 shard_rank: int = 0, shard_world: int = 1,
                 ep_active: bool = False, offload: bool = False):
        if isinstance(model_or_groups, nn.Module):
            groups = split_decay_groups(model_or_groups, ep_active=ep_active)
        else:
            groups = list(model_or_groups)
        self.defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay)
        self.betas = betas
        self.eps = eps
        self.max_grad_norm = max_grad_norm
        self.shard_rank = shard_rank
        self.shard_world = shard_world
        self.ep_active = ep_active
        self.offload = offload
        self.step_count = 0
        # expert groups are EP-sharded by construction -> no ZeRO shard on top
        self.groups: List[_FlatGroup] = [
            _FlatGroup(list(g["params"]), lr=g.get("lr", lr),
                       weight_decay=(weight_decay if g.get("weight_decay") is None
                                     else g["weight_decay"]),
                       shard_rank=0 if g.get("comm") in ("expert", "tp")
                       else shard_rank,
                       shard_world=1 if g.get("comm") in ("expert", "tp")
                       else shard_world,
                       comm=g.get("comm", "dp"), offload=offload)
            for g in groups
        ]
        self._last_norm_sq: Optional[torch.Tensor] = None
        self._norm_pinned: Optional[torch.Tensor] = None
        self._norm_event = None
        self._norm_cached = 0.0

    # ---- torch-compat surface ---------------------------------------------
    @property
    def param_groups(self):
        return [_GroupProxy(g) for g in self.groups]

    def zero_grad(self, set_to_none: bool = False):
        for g in self.groups:
            g.flat_g.zero_()

    def local_grad_norm_sq(self, shard_only: bool = False) -> torch.Tensor:
        total = None
        for g in self.groups:
            t = g.update_grad() if shard_only else g.flat_g
            ns = K.l2norm_sq(t)
            total = ns if total is None else total + ns
        return total

    @torch.no_grad()
    def step(self, grad_scale: float = 1.0, closure=None,
             norm_sq: Optional[torch.Tensor] = None,
             shard_only: bool = False):
        """Fused clip + AdamW.

        grad_scale multiplies stored grads (1/accum, DP averaging, ...).
        norm_sq: externally supplied global ||g||^2 (ZeRO engines pass the
        all-reduced value); default computes it locally.
        shard_only: grads already reduced into this rank's shard (ZeRO-2).
        """
        from ..ops.interface import invalidate_pad_cache
        invalidate_pad_cache()    # weights change: padded-K grad_x cache
        self.step_count += 1
        if norm_sq is None and self.max_grad_norm > 0:
            norm_sq = self.local_grad_norm_sq(shard_only=shard_only)
        self._last_norm_sq = norm_sq
        if norm_sq is not None and norm_sq.is_cuda:
            # async D2H so last_grad_norm() never syncs the step pipeline
            if self._norm_pinned is None:
                self._norm_pinned = torch.zeros(1, pin_memory=True)
                self._norm_event = torch.cuda.Event()
            self._norm_pinned.copy_(norm_sq.sum().reshape(1).float(),
                                    non_blocking=True)
            self._norm_event.record()
        offloaded = [g for g in self.groups
                     if g.offload and g.device.type == "cuda"]
        for g in self.groups:
            if g in offloaded:
                continue
            K.adamw_step(
                g.master, g.update_grad(), g.m, g.v, g.update_weight_out(),
                g.lr, self.betas[0], self.betas[1], self.eps, g.weight_decay,
                self.step_count, norm_sq, self.max_grad_norm, grad_scale)
        if offloaded:
            self._step_offloaded(offloaded, norm_sq, grad_scale)

    @torch.no_grad()
    def _step_offloaded(self, groups: List["_FlatGroup"],
                        norm_sq: Optional[torch.Tensor], grad_scale: float):
        """CPU AdamW on host-resident state: async D2H grad staging, one
        sync, vectorized host math, async H2D of the refreshed low-precision
        weight shard. GPU-resident groups' kernels were launched first so
        they overlap with the transfers."""
        for g in groups:
            g._g_stage.copy_(g.update_grad(), non_blocking=True)
        torch.cuda.synchronize()
        norm_cpu = (norm_sq.sum().cpu() if norm_sq is not None else None)
        if norm_cpu is not None:
            gn = float(norm_cpu) ** 0.5 * grad_scale
            if not math.isfinite(gn):
                return  # NaN/Inf grads: skip, and keep flat_w untouched
        for g in groups:
            K.adamw_step(
                g.master, g._g_stage, g.m, g.v, g._w_stage,
                g.lr, self.betas[0], self.betas[1], self.eps, g.weight_decay,
                self.step_count, norm_cpu, self.max_grad_norm, grad_scale)
            g.flat_w[g.shard_lo:g.shard_hi].copy_(g._w_stage,
                                                  non_blocking=True)

    def last_grad_norm(self) -> float:
        """Host-visible grad norm of the most recent COMPLETED step. On GPU
        this is non-blocking (value lags one step if the copy is still in
        flight); on CPU it reads directly."""
        if self._last_norm_sq is None:
            return 0.0
        if self._last_norm_sq.is_cuda:
            if self._norm_event is not None and self._norm_event.query():
                val = float(self._norm_pinned[0])
                self._norm_cached = (math.sqrt(val)
                                     if math.isfinite(val) and val >= 0
                                     else float("nan"))
            return self._norm_cached
        val = float(self._last_norm_sq.sum())
        return math.sqrt(val) if math.isfinite(val) and val >= 0 else float("nan")

    # ---- state dict --------------------------------------------------------
    def state_dict(self) -> Dict:
        return {
            "step_count": self.step_count,
            "defaults": self.defaults,
            "shard_rank": self.shard_rank,
            "shard_world": self.shard_world,
            "groups": [
                {"master": g.master.cpu(), "m": g.m.cpu(), "v": g.v.cpu(),
                 "lr": g.lr, "weight_decay": g.weight_decay, "numel": g.numel}
                for g in self.groups
            ],
        }

    def load_state_dict(self, sd: Dict):
        self.step_count = sd.get("step_count", 0)
        if sd.get("shard_world", 1) != self.shard_world:
            raise ValueError("FlatAdamW: world size changed — collect every "
                             "saved rank's shard and call load_resharded() "
                             f"(saved world={sd.get('shard_world')}, "
                             f"current={self.shard_world})")
        for g, gs in zip(self.groups, sd["groups"]):
            if g.numel != gs["numel"]:
                raise ValueError(f"FlatAdamW state size mismatch ({g.numel} vs "
                                 f"{gs['numel']})")
            g.master.copy_(gs["master"].to(g.master.device))
            g.m.copy_(gs["m"].to(g.m.device))
            g.v.copy_(gs["v"].to(g.v.device))
            g.lr = gs.get("lr", g.lr)
            g.weight_decay = gs.get("weight_decay", g.weight_decay)
            if not g._master_is_params:
                g.flat_w[g.shard_lo:g.shard_hi].copy_(g.master.to(g.dtype))

    def load_resharded(self, shard_sds: List[Dict]):
        """Elastic resume: rebuild this rank's optimizer state from the
        per-rank shards of a run saved at a DIFFERENT world size. shard_sds
        is the saved ranks' state dicts in rank order (rank 0's lives in the
        main checkpoint payload, the rest in optim_shard_*_rank{r}.pt
        siblings). Sharding is contiguous slicing of the flat buffer, so
        concatenating the saved shards in rank order reconstructs the full
        padded-at-world-A buffer; the first `numel` elements are re-padded
        for world B and re-sliced."""
        if any(g.comm in ("expert", "tp") for g in self.groups):
            raise ValueError("elastic resharding with expert/tensor "
                             "parallelism is not supported (shard placement "
                             "changes)")
        saved_world = shard_sds[0].get("shard_world", 1)
        if len(shard_sds) != saved_world:
            raise ValueError(f"need all {saved_world} shards, got "
                             f"{len(shard_sds)}")
        self.step_count = shard_sds[0].get("step_count", 0)
        for gi, g in enumerate(self.groups):
            saved_numel = shard_sds[0]["groups"][gi]["numel"]
            if saved_numel != g.numel:
                raise ValueError(f"group {gi} size mismatch ({g.numel} vs "
                                 f"{saved_numel})")
            for key, dst in (("master", g.master), ("m", g.m), ("v", g.v)):
                full = torch.cat([sd["groups"][gi][key].float()
                                  for sd in shard_sds])
                buf = torch.zeros(g.padded, dtype=torch.float32)
                buf[:g.numel] = full[:g.numel]
                if g._master_is_params:
                    dst.copy_(buf.to(dst.device))
                else:
                    dst.copy_(buf[g.shard_lo:g.shard_hi].to(dst.device))
            g.lr = shard_sds[0]["groups"][gi].get("lr", g.lr)
            g.weight_decay = shard_sds[0]["groups"][gi].get(
                "weight_decay", g.weight_decay)
            if not g._master_is_params:
                g.flat_w[g.shard_lo:g.shard_hi].copy_(g.master.to(g.dtype))

    # ---- dynamic architecture (expert add/prune) ---------------------------
    def rebuild(self, model: nn.Module):
        """Re-flatten after parameters changed (expert add/prune). State of
        surviving parameters is preserved by identity match (single-rank)."""
        old_state = {}
        for g in self.groups:
            if g.shard_world > 1:
                old_state = {}
                break
            for p, (off, n) in zip(g.params, g.offsets):
                old_state[id(p)] = (g.master[off:off + n].clone(),
                                    g.m[off:off + n].clone(),
                                    g.v[off:off + n].clone())
        lr = self.groups[0].lr if self.groups else self.defaults["lr"]
        groups = split_decay_groups(model, ep_active=self.ep_active)
        new_groups = []
        for gd in groups:
            wd = gd.get("weight_decay")
            expert = gd.get("comm") in ("expert", "tp")
            fg = _FlatGroup(list(gd["params"]), lr=lr,
                            weight_decay=(self.defaults["weight_decay"]
                                          if wd is None else wd),
                            shard_rank=0 if expert else self.shard_rank,
                            shard_world=1 if expert else self.shard_world,
                            comm=gd.get("comm", "dp"), offload=self.offload)
            if fg.shard_world == 1:
                for p, (off, n) in zip(fg.params, fg.offsets):
                    st = old_state.get(id(p))
                    if st is not None and st[0].numel() == n:
                        fg.master[off:off + n].copy_(st[0])
                        fg.m[off:off + n].copy_(st[1])
                        fg.v[off:off + n].copy_(st[2])
                        if not fg._master_is_params:
                            fg.flat_w[off:off + n].copy_(st[0].to(fg.dtype))
            new_groups.append(fg)
        self.groups = new_groups


class _GroupProxy(dict):
    """dict-like view letting generic code mutate group lr / weight_decay."""

    def __init__(self, group: _FlatGroup):
        super().__init__(lr=group.lr, weight_decay=group.weight_decay,
                         params=group.params)
        self._g = group

    def __setitem__(self, key, value):
        super().__setitem__(key, value)
        if key == "lr":
            self._g.lr = value
        elif key == "weight_decay":
            self._g.weight_decay = value
