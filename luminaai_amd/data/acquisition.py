"""Dataset acquisition & preparation.

Rebuild of the reference's data-acquisition CLIs
(/root/reference/Src/Main_Scripts/Dataset_download.py:49-485 — OASST
conversation-tree extraction, quality filtering, size-limited JSONL shards;
multi_source_dataset.py:277-1350 — per-source text cleaning processors).

Downloads require network access (HF datasets / requests). This module
separates ACQUISITION (network, optional) from PREPARATION (local, always
available): every prepare_* function also accepts already-downloaded local
files, so the pipeline is fully usable offline.
"""

from __future__ import annotations

import json
import os
import re
from typing import Callable, Dict, Iterable, List, Optional

from ..utils.data_processing import flatten_conversation_tree


def network_available() -> bool:
    try:
        import socket
        socket.create_connection(("huggingface.co", 443), timeout=3).close()
        return True
    except OSError:
        return False


# ---------------------------------------------------------------- quality
def conversation_quality(conv: Dict) -> float:
    """0..1 heuristic score (reference Dataset_download.py quality filter:
    length, turn balance, non-degenerate content)."""
    msgs = conv.get("messages", [])
    if len(msgs) < 2:
        return 0.0
    score = 1.0
    total_chars = sum(len(str(m.get("content", ""))) for m in msgs)
    if total_chars < 40:
        score *= 0.3
    if total_chars > 20000:
        score *= 0.7
    roles = [m.get("role") for m in msgs]
    if "assistant" not in roles and "ai" not in roles:
        score *= 0.2
    for m in msgs:
        c = str(m.get("content", ""))
        if c and len(set(c)) < max(3, len(c) // 50):  # degenerate repetition
            score *= 0.2
    return score


def shard_jsonl(rows: Iterable[Dict], out_dir: str, prefix: str = "shard",
                max_bytes: int = 256 * 1024 * 1024) -> List[str]:
    """Write rows into size-limited JSONL shards
    (reference Dataset_download.py:380-485)."""
    os.makedirs(out_dir, exist_ok=True)
    paths, cur, size = [], None, 0
    idx = 0
    f = None
    for row in rows:
        line = json.dumps(row, ensure_ascii=False) + "\n"
        b = len(line.encode())
        if f is None or size + b > max_bytes:
            if f:
                f.close()
            cur = os.path.join(out_dir, f"{prefix}_{idx:04d}.jsonl")
            f = open(cur, "w")
            paths.append(cur)
            idx += 1
            size = 0
        f.write(line)
        size += b
    if f:
        f.close()
    return paths


# ---------------------------------------------------------------- OASST
def prepare_oasst(input_path: str, out_dir: str,
                  min_quality: float = 0.5,
                  max_bytes: int = 256 * 1024 * 1024) -> Dict:
    """Local OASST-format trees (.jsonl of message trees) -> filtered linear
    conversations, sharded. Returns stats."""
    convs: List[Dict] = []
    with open(input_path, encoding="utf-8") as f:
        for line in f:
            line = line.strip()
            if not line:
                continue
            try:
                tree = json.loads(line)
            except json.JSONDecodeError:
                continue
            convs.extend(flatten_conversation_tree(tree))
    kept = [c for c in convs if conversation_quality(c) >= min_quality]
    paths = shard_jsonl(kept, out_dir, "oasst", max_bytes)
    return {"input_conversations": len(convs), "kept": len(kept),
            "shards": paths}


def download_oasst(out_dir: str, **kw) -> Dict:
    """Network path: fetch OASST1 via HF datasets then prepare. Raises a
    clear error offline (reference Dataset_download.py:49-120)."""
    if not network_available():
        raise RuntimeError(
            "no network access — download OASST elsewhere and run "
            "prepare_oasst() on the local file")
    from datasets import load_dataset  # noqa: WPS433
    ds = load_dataset("OpenAssistant/oasst1", split="train")
    tmp = os.path.join(out_dir, "_oasst_raw.jsonl")
    os.makedirs(out_dir, exist_ok=True)
    with open(tmp, "w") as f:
        for row in ds:
            f.write(json.dumps(dict(row)) + "\n")
    return prepare_oasst(tmp, out_dir, **kw)


# ---------------------------------------------------------- text sources
_WS = re.compile(r"[ \t]+")
_MULTI_NL = re.compile(r"\n{3,}")


def _clean_generic(text: str) -> str:
    text = _WS.sub(" ", text)
    text = _MULTI_NL.sub("\n\n", text)
    return text.strip()


def _clean_wikipedia(text: str) -> str:
    text = re.sub(r"==+ *(References|External links|See also)"
                  r" *==+.*", "", text, flags=re.S)
    text = re.sub(r"==+ *([^=]+?) *==+", r"\1.", text)
    return _clean_generic(text)


def _clean_code_qa(text: str) -> str:
    text = re.sub(r"<[^>]+>", "", text)      # html tags
    return _clean_generic(text)


SOURCE_PROCESSORS: Dict[str, Callable[[str], str]] = {
    "wikipedia": _clean_wikipedia,
    "gutenberg": _clean_generic,
    "arxiv": _clean_generic,
    "stackoverflow": _clean_code_qa,
    "pubmed": _clean_generic,
    "openwebtext": _clean_generic,
    "philpapers": _clean_generic,
    "cc_news": _clean_generic,
}


def prepare_text_corpus(input_path: str, out_path: str, source: str = "generic",
                        min_chars: int = 200) -> Dict:
    """Clean a local raw-text/JSONL dump into a training .txt corpus
    (reference multi_source_dataset.py per-source processors)."""
    proc = SOURCE_PROCESSORS.get(source, _clean_generic)
    n_in = n_out = 0
    os.makedirs(os.path.dirname(out_path) or ".", exist_ok=True)
    with open(input_path, encoding="utf-8", errors="replace") as fin, \
            open(out_path, "w") as fout:
        for line in fin:
            n_in += 1
            text = line
            if line.lstrip().startswith("{"):
                try:
                    row = json.loads(line)
                    text = row.get("text") or row.get("content") or ""
                except json.JSONDecodeError:
                    pass
            text = proc(text)
            if len(text) >= min_chars:
                fout.write(text + "\n")
                n_out += 1
    return {"lines_in": n_in, "kept": n_out, "source": source,
            "out_path": out_path}


def main(argv=None):
    """CLI: prepare local corpora (reference Dataset_download.py /
    multi_source_dataset.py CLIs, network-optional)."""
    import argparse
    ap = argparse.ArgumentParser(prog="luminaai-amd data",
                                 description="dataset acquisition/preparation")
    sub = ap.add_subparsers(dest="cmd", required=True)
    p1 = sub.add_parser("oasst", help="prepare OASST message trees")
    p1.add_argument("input", help="local trees .jsonl (or 'download')")
    p1.add_argument("--out", default="data/oasst")
    p1.add_argument("--min-quality", type=float, default=0.5)
    p2 = sub.add_parser("text", help="clean a raw text/JSONL corpus")
    p2.add_argument("input")
    p2.add_argument("--out", default="data/corpus.txt")
    p2.add_argument("--source", default="generic",
                    choices=sorted(SOURCE_PROCESSORS) + ["generic"])
    p3 = sub.add_parser("validate", help="validate a conversations JSONL")
    p3.add_argument("input")
    args = ap.parse_args(argv)
    if args.cmd == "oasst":
        stats = (download_oasst(args.out, min_quality=args.min_quality)
                 if args.input == "download"
                 else prepare_oasst(args.input, args.out,
                                    min_quality=args.min_quality))
    elif args.cmd == "text":
        stats = prepare_text_corpus(args.input, args.out, source=args.source)
    else:
        from ..utils.data_processing import validate_jsonl
        stats = validate_jsonl(args.input)
    print(json.dumps(stats, indent=2, default=str))
    return stats


if __name__ == "__main__":
    main()
