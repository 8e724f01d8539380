"""Native data-parallel / ZeRO engine on RCCL over xGMI.

One engine replaces the reference's DeepSpeed/FSDP/ColossalAI backends
(reference backend_deepspeed.py:129-165 ZeRO config, backend_fsdp.py:151-198
sharding strategies, vendored low_level_optim.py:370-681). Built directly on
the FlatAdamW flat buffers:

  stage 0 (DDP):    bucketed all-reduce of the flat grad, overlapped with
                    backward via post-accumulate-grad hooks.
  stage 1:          same grad all-reduce; optimizer state + update sharded
                    (master/m/v are shard-size in FlatAdamW); bf16 weight
                    all-gather after the update.
  stage 2:          reduce-scatter of the flat grad into this rank's shard
                    (half the wire bytes of all-reduce on xGMI), sharded
                    update, bf16 weight all-gather.
  stage 3:          parameter sharding lives in its own engine
                    (parallel/zero3.py: block-unit resizable flat buffers,
                    gather/free hooks, shard AdamW); the Trainer selects it
                    for config.zero_stage >= 3.

Bucket size defaults to ~50 MB of grad per collective: xGMI is 7 independent
point-to-point links, so several in-flight medium buckets keep all links busy
where one 1 GB ring transfer is single-link bound.
"""

from __future__ import annotations

import math
from typing import List, Optional

from typing import TYPE_CHECKING

import torch
import torch.distributed as dist

from . import comm
from ..utils.profiling import profiling_context

if TYPE_CHECKING:  # avoid circular import at runtime
    from ..training.optimizer import FlatAdamW


class ZeroEngine:
    """Wraps a FlatAdamW whose flat buffers this engine reduces/gathers."""

    def __init__(self, optimizer: "FlatAdamW", stage: int = 0,
                 bucket_bytes: int = 50_000_000,
                 overlap_comm: bool = True,
                 process_group=None, mesh=None):
        assert stage in (0, 1, 2), "use parallel/zero3.Zero3Engine for stage 3"
        self.opt = optimizer
        self.stage = stage
        self.bucket_bytes = bucket_bytes
        self.overlap = overlap_comm
        self.pg = process_group
        self.mesh = mesh                    # ParallelMesh when EP is active
        self.world = comm.get_world_size()
        self.rank = comm.get_rank()
        self._hooks = []
        self._works: List = []
        self._bucket_plan = None
        if stage in (1, 2):
            assert optimizer.shard_world == self.world, (
                "FlatAdamW must be built with shard_world=world for ZeRO-1/2")
        if self.world > 1 and self.overlap and stage in (0, 1):
            self._install_hooks()

    def _dp_groups(self):
        return [g for g in self.opt.groups if g.comm == "dp"]

    def _expert_groups(self):
        return [g for g in self.opt.groups if g.comm in ("expert", "tp")]

    # ------------------------------------------------------------- hooks
    def _install_hooks(self):
        """Bucketed all-reduce launched as grads become ready during backward.
        Buckets are contiguous ranges of the flat grad buffer; a bucket fires
        when every param mapping into it has accumulated its grad."""
        self.sync_enabled = True
        plan = []  # (group, lo, hi, param_ids)
        for g in self._dp_groups():
            lo = 0
            cur_ids = []
            cur_hi = 0
            for p, (off, n) in zip(g.params, g.offsets):
                cur_ids.append(id(p))
                cur_hi = off + n
                if (cur_hi - lo) * g.flat_g.element_size() >= self.bucket_bytes:
                    plan.append((g, lo, cur_hi, set(cur_ids)))
                    lo, cur_ids = cur_hi, []
            if cur_ids or g.padded > lo:
                plan.append((g, lo, g.padded, set(cur_ids)))
        self._bucket_plan = plan
        self._pending = {}

        def make_hook(bucket_idx):
            def hook(param):
                if not self.sync_enabled:
                    return
                g, lo, hi, ids = self._bucket_plan[bucket_idx]
                rem = self._pending[bucket_idx]
                rem.discard(id(param))
                if not rem:
                    w = dist.all_reduce(g.flat_g[lo:hi], async_op=True,
                                        group=self.pg)
                    self._works.append(w)
            return hook

        param_to_bucket = {}
        for bi, (g, lo, hi, ids) in enumerate(plan):
            for pid in ids:
                param_to_bucket[pid] = bi
        for g in self._dp_groups():
            for p in g.params:
                h = p.register_post_accumulate_grad_hook(
                    make_hook(param_to_bucket[id(p)]))
                self._hooks.append(h)
        self.reset_bucket_state()

    def reset_bucket_state(self):
        if self._bucket_plan is not None:
            self._pending = {bi: set(ids) for bi, (g, lo, hi, ids)
                             in enumerate(self._bucket_plan)}

    def set_sync(self, enabled: bool):
        """Disable collective launches during gradient-accumulation
        micro-steps; enable for the boundary micro-step."""
        self.sync_enabled = enabled if self._bucket_plan is not None else False

    # ------------------------------------------------------------- step
    def reduce_gradients(self):
        with profiling_context("comm.reduce_gradients"):
            return self._reduce_gradients_impl()

    def _reduce_gradients_impl(self):
        """Complete (or launch) the gradient reduction for this step."""
        if self.world <= 1:
            return
        if self.stage in (0, 1):
            if self._bucket_plan is not None and self.sync_enabled:
                # buckets whose params never produced a grad this step (e.g.
                # a MoD router disabled at capacity 1.0) never fired their
                # hook — reduce them now or the ranks silently diverge
                for bi, rem in self._pending.items():
                    if rem:
                        g, lo, hi, _ = self._bucket_plan[bi]
                        self._works.append(
                            dist.all_reduce(g.flat_g[lo:hi], async_op=True,
                                            group=self.pg))
                for w in self._works:
                    w.wait()
                self._works.clear()
                self.reset_bucket_state()
            else:
                for g in self._dp_groups():
                    dist.all_reduce(g.flat_g, group=self.pg)
        else:  # stage 2: reduce-scatter into this rank's shard
            for g in self._dp_groups():
                shard = g.flat_g[g.shard_lo:g.shard_hi]
                if self._backend() == "gloo":
                    # gloo lacks reduce_scatter; tests-only fallback
                    dist.all_reduce(g.flat_g, group=self.pg)
                else:
                    tmp = torch.empty_like(shard)
                    dist.reduce_scatter_tensor(tmp, g.flat_g, group=self.pg)
                    shard.copy_(tmp)
        # sharded-param grads (EP expert shards / TP weight shards):
        # replicas of the same shard live across the replica group (no-op
        # when every replica set has one member)
        if self.mesh is not None:
            for g in self._expert_groups():
                if self.mesh.replica_size_for(g.comm) > 1:
                    dist.all_reduce(
                        g.flat_g, group=self.mesh.replica_group_for(g.comm))
        if self.mesh is not None and self.mesh.tp_size > 1:
            # TP ranks SHARE their batch: dense grads sum tp duplicates and
            # the uniform 1/world grad_scale averages them out, but a weight
            # shard's grad is computed once — pre-scale by tp so the same
            # grad_scale (and the same clip norm) applies to every group
            for g in self._expert_groups():
                g.flat_g.mul_(float(self.mesh.tp_size))

    def global_grad_norm_sq(self) -> Optional[torch.Tensor]:
        if self.opt.max_grad_norm <= 0:
            return None
        from ..ops import interface as K
        shard_only = self.stage == 2
        ns = None
        for g in self._dp_groups():
            t = g.update_grad() if shard_only else g.flat_g
            n = K.l2norm_sq(t)
            ns = n if ns is None else ns + n
        if ns is not None and self.world > 1 and shard_only:
            dist.all_reduce(ns, group=self.pg)
        # stages 0/1: dp grads are already globally reduced (summed); the norm
        # of the summed grad is what clipping applies to (after grad_scale).
        for g in self._expert_groups():
            ns_e = K.l2norm_sq(g.flat_g)
            if self.mesh is not None \
                    and self.mesh.exchange_size_for(g.comm) > 1:
                # each rank holds 1/N of the sharded params (post replica
                # reduce) -> summing over ONE exchange group covers every
                # shard exactly once
                dist.all_reduce(ns_e,
                                group=self.mesh.exchange_group_for(g.comm))
            ns = ns_e if ns is None else ns + ns_e
        return ns

    def step(self, grad_scale: float = 1.0):
        with profiling_context("comm.zero_step"):
            return self._step_impl(grad_scale)

    def _step_impl(self, grad_scale: float = 1.0):
        """reduce -> (norm) -> sharded fused update -> weight all-gather."""
        self.reduce_gradients()
        norm_sq = self.global_grad_norm_sq()
        self.opt.step(grad_scale=grad_scale, norm_sq=norm_sq,
                      shard_only=self.stage == 2)
        if self.stage in (1, 2) and self.world > 1:
            for g in self._dp_groups():
                shard = g.update_weight_out()
                if self._backend() == "gloo":
                    chunks = list(g.weight_view().chunk(self.world))
                    dist.all_gather(chunks, shard.contiguous(), group=self.pg)
                else:
                    dist.all_gather_into_tensor(g.weight_view(), shard,
                                                group=self.pg)

    def zero_grad(self):
        self.opt.zero_grad()
        self.reset_bucket_state()
        self._works.clear()

    def _backend(self) -> str:
        try:
            return dist.get_backend(self.pg)
        except RuntimeError:
            return "gloo"

    # ------------------------------------------------------------- misc
    def broadcast_parameters(self):
        """Initial weight sync: replicated (dp) params from rank 0 to all;
        EP-sharded expert params from the first rank of each expert-dp
        replica set (they differ across EP ranks by design)."""
        if self.world <= 1:
            return
        for g in self._dp_groups():
            dist.broadcast(g.weight_view(), src=0, group=self.pg)
            if not g._master_is_params:
                g.master.copy_(
                    g.weight_view()[g.shard_lo:g.shard_hi].float())
        if self.mesh is not None:
            for g in self._expert_groups():
                if self.mesh.replica_size_for(g.comm) <= 1:
                    continue
                pg = self.mesh.replica_group_for(g.comm)
                src = dist.get_process_group_ranks(pg)[0]
                dist.broadcast(g.weight_view(), src=src, group=pg)
                if not g._master_is_params:
                    g.master.copy_(g.weight_view().float())

    def remove_hooks(self):
        for h in self._hooks:
            h.remove()
        self._hooks.clear()


def create_engine(optimizer: "FlatAdamW", config) -> ZeroEngine:
    return ZeroEngine(optimizer, stage=min(config.zero_stage, 2),
                      bucket_bytes=config.reduce_bucket_size * 2
                      if config.reduce_bucket_size < 10_000_000 else
                      config.reduce_bucket_size,
                      overlap_comm=config.overlap_comm)
