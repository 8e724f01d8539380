"""Checkpoint discovery and smart loading for inference.

Rebuild of the reference loaders (/root/reference/Src/Main_Scripts/Chat.py:
load_checkpoint_smart :132 (prefix stripping), load_zero_shards :163
(ZeRO-shard merge), infer_config_from_state_dict :219 (architecture from
tensor shapes), find_latest_checkpoint :301)."""

from __future__ import annotations

import glob
import os
import re
from typing import Dict, List, Optional

import torch

from ..models.transformer import DeepSeekConfig

_STRIP_PREFIXES = ("module.", "_orig_mod.", "model.")


def _strip_prefixes(sd: Dict[str, torch.Tensor]) -> Dict[str, torch.Tensor]:
    out = {}
    for k, v in sd.items():
        for p in _STRIP_PREFIXES:
            if k.startswith(p):
                k = k[len(p):]
        out[k] = v
    return out


def load_checkpoint_smart(path: str, map_location="cpu") -> Dict:
    """Load a checkpoint file; returns {"model_state_dict": ..., ...} with
    wrapper prefixes stripped. Accepts raw state dicts too."""
    payload = torch.load(path, map_location=map_location, weights_only=False)
    if isinstance(payload, dict) and "stage_state_dict" in payload:
        # pipeline-parallel per-stage save: merge all sibling stages
        return merge_pp_checkpoints(find_pp_stages(path),
                                    map_location=map_location)
    if isinstance(payload, dict) and "model_state_dict" in payload:
        payload["model_state_dict"] = _strip_prefixes(payload["model_state_dict"])
        return payload
    if isinstance(payload, dict) and all(
            torch.is_tensor(v) for v in payload.values()):
        return {"model_state_dict": _strip_prefixes(payload)}
    raise ValueError(f"unrecognised checkpoint format in {path}")


def load_zero_shards(shard_dir: str, map_location="cpu") -> Dict:
    """Merge per-rank ZeRO checkpoint shards (rank{N}.pt / *_rank_{N}.pt)
    into one full state dict. Shards carry disjoint flat-optimizer partitions;
    the model_state_dict is identical across ranks (stage 1/2) so the merge
    takes rank0's model and concatenates optimizer shards per group."""
    patterns = ["*rank*[0-9].pt", "*shard*[0-9].pt"]
    files: List[str] = []
    for p in patterns:
        files.extend(glob.glob(os.path.join(shard_dir, p)))
    files = sorted(set(files),
                   key=lambda f: int(re.findall(r"(\d+)", os.path.basename(f))[-1]))
    if not files:
        raise FileNotFoundError(f"no shard files in {shard_dir}")
    shards = [torch.load(f, map_location=map_location, weights_only=False)
              for f in files]
    merged = dict(shards[0])
    merged["model_state_dict"] = _strip_prefixes(shards[0]["model_state_dict"])
    opt_shards = [s.get("optimizer_state_dict") for s in shards]
    if all(o is not None for o in opt_shards) and \
            all("groups" in o for o in opt_shards):
        groups = []
        for gi in range(len(opt_shards[0]["groups"])):
            g0 = dict(opt_shards[0]["groups"][gi])
            for key in ("exp_avg", "exp_avg_sq", "master"):
                parts = [o["groups"][gi][key] for o in opt_shards
                         if o["groups"][gi].get(key) is not None]
                if parts:
                    g0[key] = torch.cat(parts)
            groups.append(g0)
        merged["optimizer_state_dict"] = dict(opt_shards[0], groups=groups)
    return merged


def merge_ep_checkpoints(paths: List[str], map_location="cpu") -> Dict:
    """Merge per-EP-rank checkpoint files (written by Trainer.save_checkpoint
    under expert parallelism as *_ep_rank_{r}.pt) into one full-expert
    payload: expert weights (w_gate_up / w_down) are concatenated along the
    expert dim in ep-rank order; everything else is taken from rank 0."""
    def _ep_rank(p):
        m = re.search(r"_ep_rank_(\d+)", os.path.basename(p))
        return int(m.group(1)) if m else 0

    paths = sorted(paths, key=_ep_rank)
    payloads = [torch.load(p, map_location=map_location, weights_only=False)
                for p in paths]
    sds = [_strip_prefixes(p["model_state_dict"]) for p in payloads]
    merged_sd = dict(sds[0])
    for key in sds[0]:
        if ".w_gate_up" in key or ".w_down" in key:
            merged_sd[key] = torch.cat([sd[key] for sd in sds], dim=0)
    out = dict(payloads[0])
    out["model_state_dict"] = merged_sd
    return out


def merge_pp_checkpoints(paths: List[str], map_location="cpu") -> Dict:
    """Reassemble per-stage pipeline checkpoints (pp_stage_rank{r}.pt from
    training/pipeline_loop.py) into one full model_state_dict. Handles the
    1F1B layout (one PipelineStage per rank) and interleaved virtual
    stages (a ModuleList of chunks per rank, global stage s = c*pp + r);
    local layer indices are re-based onto the global layer numbering."""
    def _rank(p):
        m = re.search(r"pp_stage_rank(\d+)", os.path.basename(p))
        return int(m.group(1)) if m else 0

    paths = sorted(paths, key=_rank)
    payloads = [torch.load(p, map_location=map_location, weights_only=False)
                for p in paths]
    pp = payloads[0].get("pp_world", len(payloads))
    v = payloads[0].get("virtual_stages", 1)
    if len(payloads) != pp:
        raise ValueError(f"need all {pp} stage files, got {len(payloads)}")

    # chunk (global stage) -> its layer key prefix and state source
    chunks: List[Dict] = [None] * (pp * v)
    for r, pl in enumerate(payloads):
        sd = pl["stage_state_dict"]
        if v == 1:
            chunks[r] = sd
        else:
            for c in range(v):
                pref = f"{c}."
                chunks[c * pp + r] = {k[len(pref):]: t for k, t in sd.items()
                                      if k.startswith(pref)}
    merged: Dict[str, torch.Tensor] = {}
    layer_off = 0
    for s, sd in enumerate(chunks):
        n_local = 0
        for k, t in sd.items():
            if k.startswith("layers."):
                rest = k.split(".", 2)
                li = int(rest[1])
                n_local = max(n_local, li + 1)
                merged[f"layers.{layer_off + li}.{rest[2]}"] = t
            else:   # embed (stage 0), final_norm/lm_head (last stage)
                merged[k] = t
        layer_off += n_local
    return {"model_state_dict": merged,
            "global_step": payloads[0].get("global_step", 0)}


def find_pp_stages(any_stage_path: str) -> List[str]:
    """All sibling pp_stage_rank* files of one pipeline checkpoint."""
    base = re.sub(r"pp_stage_rank\d+", "pp_stage_rank*", any_stage_path)
    return sorted(glob.glob(base))


def find_ep_shards(any_shard_path: str) -> List[str]:
    """All sibling _ep_rank_* files of one EP checkpoint shard."""
    base = re.sub(r"_ep_rank_\d+", "_ep_rank_*", any_shard_path)
    return sorted(glob.glob(base))


def infer_config_from_state_dict(sd: Dict[str, torch.Tensor]) -> DeepSeekConfig:
    """Reconstruct a DeepSeekConfig from tensor shapes
    (reference Chat.py:219-300)."""
    vocab_size, hidden = sd["embed_tokens.weight"].shape
    layer_ids = {int(m.group(1)) for k in sd
                 if (m := re.match(r"layers\.(\d+)\.", k))}
    num_layers = max(layer_ids) + 1 if layer_ids else 0

    qkv = sd["layers.0.attention.qkv_proj.weight"]      # [q + 2kv, h]
    o = sd["layers.0.attention.o_proj.weight"]          # [h, q]
    q_size = o.shape[1]
    kv_size = (qkv.shape[0] - q_size) // 2

    # head_dim from the RoPE-compatible assumption head_dim = hidden/num_heads;
    # num_heads is the largest divisor consistent with q_size == hidden.
    # q_size == hidden always holds for this family, so take gcd-style probe:
    num_heads = None
    for h in (64, 48, 40, 32, 24, 20, 16, 12, 8, 6, 4, 2, 1):
        if hidden % h == 0 and q_size % h == 0 and kv_size % (hidden // h) == 0:
            num_heads = h
            break
    num_heads = num_heads or 1
    head_dim = hidden // num_heads
    num_kv_heads = max(1, kv_size // head_dim)

    use_moe = any(".w_gate_up" in k for k in sd)
    num_experts = 8
    moe_inter = None
    if use_moe:
        for k, v in sd.items():
            if k.endswith(".w_gate_up"):
                num_experts, _, i2 = v.shape
                moe_inter = i2 // 2
                break
    inter = moe_inter
    for k, v in sd.items():
        if k.endswith("ffn.gate_up_proj.weight"):
            inter = v.shape[0] // 2
            break
    use_mod = any("mod_router" in k for k in sd)
    tie = "lm_head.weight" not in sd or \
        sd["lm_head.weight"].data_ptr() == sd["embed_tokens.weight"].data_ptr()

    return DeepSeekConfig(
        vocab_size=vocab_size, hidden_size=hidden, num_layers=num_layers,
        num_heads=num_heads, num_kv_heads=num_kv_heads,
        intermediate_size=inter, use_moe=use_moe, num_experts=num_experts,
        use_mod=use_mod, tie_word_embeddings=tie,
        moe_pattern="all" if use_moe and all(
            f"layers.{i}.ffn.w_gate_up" in sd for i in range(num_layers))
        else ("every_2nd" if use_moe else "all"),
    )


def find_latest_checkpoint(search_dirs: Optional[List[str]] = None) -> Optional[str]:
    """Newest .pt under checkpoints/ or experiments/ (reference Chat.py:301)."""
    search_dirs = search_dirs or ["checkpoints", "experiments", "."]
    candidates: List[str] = []
    for d in search_dirs:
        candidates.extend(glob.glob(os.path.join(d, "**", "*.pt"),
                                    recursive=True))
    candidates = [c for c in candidates if not c.endswith(".tmp")]
    if not candidates:
        return None
    return max(candidates, key=os.path.getmtime)
