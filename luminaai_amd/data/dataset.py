"""Arrow-backed data pipeline.

Rebuild of the reference dataset layer
(/root/reference/Src/Main_Scripts/core/dataset.py:47-957): HF-datasets/Arrow
memory-mapped loading, batched tokenize -> fixed-length chunking for base
(pre-training) text, per-conversation encoding with shifted inputs/labels and
per-token loss weights for fine-tuning, hybrid/interleaved combination, and a
tuned DataLoader factory. Works fully offline (local txt/jsonl/parquet files);
`datasets` is optional — a plain in-memory path covers environments without it.
"""

from __future__ import annotations

import json
import os
from pathlib import Path
from typing import Dict, Iterator, List, Optional, Sequence

import torch
from torch.utils.data import ConcatDataset, DataLoader, Dataset, IterableDataset

try:
    import datasets as hf_datasets
    _HAS_HF = True
except ImportError:
    hf_datasets = None
    _HAS_HF = False


# ----------------------------------------------------------------------
class SyntheticDataset(Dataset):
    """Random-token dataset for benchmarks (BASELINE: synthetic data)."""

    def __init__(self, vocab_size: int, seq_length: int, num_samples: int,
                 seed: int = 0):
        g = torch.Generator().manual_seed(seed)
        self.data = torch.randint(1, vocab_size, (num_samples, seq_length + 1),
                                  generator=g)

    def __len__(self):
        return self.data.shape[0]

    def __getitem__(self, i):
        row = self.data[i]
        return {
            "input_ids": row[:-1].clone(),
            "labels": row[1:].clone(),
            "loss_weights": torch.ones(row.shape[0] - 1),
        }


# ----------------------------------------------------------------------
class BaseTrainingDataset(Dataset):
    """Pre-training text -> fixed-seq_length chunks
    (reference FastBaseTrainingDataset, dataset.py:47-238)."""

    def __init__(self, path: str, tokenizer, seq_length: int,
                 cache_dir: Optional[str] = None):
        self.seq_length = seq_length
        self.tokenizer = tokenizer
        texts = _load_texts(path)
        ids: List[int] = []
        eos = tokenizer.special_tokens["<|endoftext|>"]
        for t in texts:
            ids.extend(tokenizer.encode(t, use_cache=False))
            ids.append(eos)
        n_chunks = max(0, (len(ids) - 1) // seq_length)
        self.chunks = [
            torch.tensor(ids[i * seq_length:(i + 1) * seq_length + 1])
            for i in range(n_chunks)
        ]

    def __len__(self):
        return len(self.chunks)

    def __getitem__(self, i):
        row = self.chunks[i]
        return {
            "input_ids": row[:-1].clone(),
            "labels": row[1:].clone(),
            "loss_weights": torch.ones(row.shape[0] - 1),
        }


class StreamingBaseTrainingDataset(IterableDataset):
    """Streaming variant for files too large to tokenize up front
    (reference FastStreamingBaseTrainingDataset, dataset.py:240-338)."""

    def __init__(self, path: str, tokenizer, seq_length: int):
        self.path = path
        self.tokenizer = tokenizer
        self.seq_length = seq_length

    @staticmethod
    def _shard() -> tuple:
        """(shard_id, num_shards) over DataLoader workers x DP ranks so the
        stream is partitioned, not duplicated. SP ranks of one replica count
        as ONE data rank (they slice the same batch along the sequence)."""
        import torch.distributed as dist
        from torch.utils.data import get_worker_info
        winfo = get_worker_info()
        wid, nw = (winfo.id, winfo.num_workers) if winfo else (0, 1)
        if dist.is_available() and dist.is_initialized() \
                and dist.get_world_size() > 1:
            from ..parallel.mesh import get_mesh
            mesh = get_mesh()
            # SP ranks slice one batch along the sequence; TP ranks compute
            # on one batch with sharded weights -> both count as ONE data
            # rank (both are innermost-contiguous in the mesh layout)
            shared = 1
            if mesh is not None:
                shared = max(mesh.sp_size, 1) * max(mesh.tp_size, 1)
            dp_world = dist.get_world_size() // shared
            dp_rank = dist.get_rank() // shared
        else:
            dp_rank, dp_world = 0, 1
        return dp_rank * nw + wid, dp_world * nw

    def __iter__(self) -> Iterator[Dict[str, torch.Tensor]]:
        eos = self.tokenizer.special_tokens["<|endoftext|>"]
        shard_id, num_shards = self._shard()
        buf: List[int] = []
        for ti, text in enumerate(_iter_texts(self.path)):
            if ti % num_shards != shard_id:
                continue
            buf.extend(self.tokenizer.encode(text, use_cache=False))
            buf.append(eos)
            while len(buf) > self.seq_length:
                row = torch.tensor(buf[:self.seq_length + 1])
                buf = buf[self.seq_length:]
                yield {
                    "input_ids": row[:-1].clone(),
                    "labels": row[1:].clone(),
                    "loss_weights": torch.ones(self.seq_length),
                }


class ConversationDataset(Dataset):
    """Fine-tuning conversations with per-token loss weights
    (reference FastConversationDataset, dataset.py:340-564)."""

    def __init__(self, path: str, tokenizer, seq_length: int,
                 min_messages: int = 1):
        self.tokenizer = tokenizer
        self.seq_length = seq_length
        self.conversations = [
            c for c in _load_jsonl(path)
            if isinstance(c, dict) and len(c.get("messages", [])) >= min_messages
        ]

    def __len__(self):
        return len(self.conversations)

    def __getitem__(self, i):
        ids, weights = self.tokenizer.encode_conversation(
            self.conversations[i], max_length=self.seq_length + 1,
            return_loss_weights=True)
        pad = self.tokenizer.pad_token_id
        L = self.seq_length + 1
        if len(ids) < L:
            weights = weights + [0.0] * (L - len(ids))
            ids = ids + [pad] * (L - len(ids))
        ids_t = torch.tensor(ids[:L])
        w_t = torch.tensor(weights[:L], dtype=torch.float32)
        labels = ids_t[1:].clone()
        labels[w_t[1:] == 0.0] = -100  # specials/pads excluded from the loss
        return {
            "input_ids": ids_t[:-1].clone(),
            "labels": labels,
            "loss_weights": w_t[1:].clone(),
        }


class InterleavedDataset(Dataset):
    """Ratio-based interleave of two datasets (reference dataset.py:762-806)."""

    def __init__(self, a: Dataset, b: Dataset, ratio_a: float = 0.5):
        self.a, self.b = a, b
        self.ratio_a = ratio_a
        self.length = len(a) + len(b)

    def __len__(self):
        return self.length

    def __getitem__(self, i):
        pick_a = (i * self.ratio_a) % 1.0 < self.ratio_a
        if pick_a and len(self.a):
            return self.a[int(i * self.ratio_a) % len(self.a)]
        if len(self.b) == 0:
            return self.a[i % len(self.a)]
        return self.b[i % len(self.b)]


# ----------------------------------------------------------------------
class HybridDatasetManager:
    """base_only / finetuning_only / hybrid / interleaved mode detection
    (reference FastHybridDatasetManager, dataset.py:566-760)."""

    def __init__(self, config, tokenizer):
        self.config = config
        self.tokenizer = tokenizer

    def _build_for(self, path: str) -> Optional[Dataset]:
        if not path or not os.path.exists(path):
            return None
        size_gb = os.path.getsize(path) / 1e9 if os.path.isfile(path) else 0.0
        if _looks_like_conversations(path):
            return ConversationDataset(path, self.tokenizer, self.config.seq_length)
        if size_gb > self.config.streaming_threshold_gb:
            return StreamingBaseTrainingDataset(path, self.tokenizer,
                                                self.config.seq_length)
        return BaseTrainingDataset(path, self.tokenizer, self.config.seq_length)

    def get_datasets(self):
        train = self._build_for(self.config.train_data_path)
        evalds = self._build_for(self.config.eval_data_path)
        if train is None:
            train = SyntheticDataset(self.config.vocab_size,
                                     self.config.seq_length, 256,
                                     seed=self.config.seed)
        return train, evalds


# ----------------------------------------------------------------------
def create_dataloader(dataset: Dataset, config, shuffle: bool = True) -> DataLoader:
    """Tuned DataLoader (reference create_fast_dataloader, dataset.py:807-845).

    Under data parallelism each rank gets a disjoint 1/world slice via
    DistributedSampler (the DP group for data purposes is the world minus
    sequence parallelism: SP ranks of one replica must see the SAME batch
    and slice it along the sequence)."""
    import torch.distributed as dist
    is_iterable = isinstance(dataset, IterableDataset)
    kwargs = dict(
        batch_size=config.micro_batch_size or 1,
        num_workers=config.num_workers,
        pin_memory=config.pin_memory and torch.cuda.is_available(),
        drop_last=True,
    )
    if not is_iterable and dist.is_available() and dist.is_initialized() \
            and dist.get_world_size() > 1:
        from ..parallel.mesh import get_mesh
        mesh = get_mesh()
        shared = 1
        if mesh is not None:      # SP slices / TP shares one batch
            shared = max(mesh.sp_size, 1) * max(mesh.tp_size, 1)
        world = dist.get_world_size() // shared
        rank = dist.get_rank() // shared
        from torch.utils.data.distributed import DistributedSampler
        kwargs["sampler"] = DistributedSampler(
            dataset, num_replicas=world, rank=rank, shuffle=shuffle,
            seed=config.seed, drop_last=True)
    elif not is_iterable:
        kwargs["shuffle"] = shuffle
    if config.num_workers > 0:
        kwargs["prefetch_factor"] = config.prefetch_factor
        kwargs["persistent_workers"] = True
    return DataLoader(dataset, **kwargs)


def setup_datasets(config, tokenizer):
    """Entry point (reference setup_fast_datasets/setup_datasets,
    dataset.py:846-957)."""
    return HybridDatasetManager(config, tokenizer).get_datasets()


# ---------------------------------------------------------------- helpers
def _load_texts(path: str) -> List[str]:
    p = Path(path)
    if p.suffix == ".txt":
        return [p.read_text(errors="replace")]
    if p.suffix in (".jsonl", ".json"):
        out = []
        for obj in _load_jsonl(path):
            if isinstance(obj, dict):
                out.append(obj.get("text", json.dumps(obj)))
            else:
                out.append(str(obj))
        return out
    if p.suffix == ".parquet":
        try:
            import pyarrow.parquet as pq
            table = pq.read_table(str(p))
            col = "text" if "text" in table.column_names \
                else table.column_names[0]
            return [str(x) for x in table.column(col).to_pylist()
                    if x is not None]
        except ImportError:
            if _HAS_HF:
                ds = hf_datasets.load_dataset("parquet", data_files=str(p),
                                              split="train")
                return [r.get("text", "") for r in ds]
            raise
    raise ValueError(f"unsupported data file {path}")


def _iter_texts(path: str) -> Iterator[str]:
    p = Path(path)
    if p.suffix == ".parquet":
        # row-group streaming: memory stays bounded for >10 GB files
        import pyarrow.parquet as pq
        pf = pq.ParquetFile(str(p))
        col = "text" if "text" in pf.schema_arrow.names \
            else pf.schema_arrow.names[0]
        for batch in pf.iter_batches(columns=[col]):
            for x in batch.column(0).to_pylist():
                if x:
                    yield str(x)
    elif p.suffix == ".txt":
        with open(p, errors="replace") as f:
            for line in f:
                if line.strip():
                    yield line
    else:
        with open(p, errors="replace") as f:
            for line in f:
                line = line.strip()
                if not line:
                    continue
                try:
                    obj = json.loads(line)
                    yield obj.get("text", line) if isinstance(obj, dict) else line
                except json.JSONDecodeError:
                    yield line


def _load_jsonl(path: str) -> List:
    out = []
    with open(path, errors="replace") as f:
        for line in f:
            line = line.strip()
            if not line:
                continue
            try:
                out.append(json.loads(line))
            except json.JSONDecodeError:
                continue
    return out


def _looks_like_conversations(path: str) -> bool:
    if not path.endswith((".jsonl", ".json")):
        return False
    try:
        with open(path, errors="replace") as f:
            for line in f:
                line = line.strip()
                if line:
                    obj = json.loads(line)
                    return isinstance(obj, dict) and "messages" in obj
    except (OSError, json.JSONDecodeError):
        return False
    return False
