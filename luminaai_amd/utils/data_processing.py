"""Data validation and preprocessing helpers.

Rebuild of /root/reference/Src/Main_Scripts/utils/data_processing.py:13-273:
OASST conversation-tree flattening, comprehensive JSONL validation with
statistics, and a sample-data generator for tests/smoke runs."""

from __future__ import annotations

import json
import random
from typing import Dict, List, Optional


def flatten_conversation_tree(tree: Dict) -> List[Dict]:
    """OASST-style message tree -> list of linear conversations
    ({"messages": [{"role", "content"}, ...]}). Walks every root->leaf path."""
    out = []

    def walk(node, path):
        msg = {"role": node.get("role", "user"),
               "content": node.get("text", node.get("content", ""))}
        path = path + [msg]
        replies = node.get("replies", [])
        if not replies:
            if len(path) >= 2:
                out.append({"messages": path})
            return
        for r in replies:
            walk(r, path)

    prompt = tree.get("prompt", tree)
    walk(prompt, [])
    return out


def validate_jsonl(path: str, max_errors: int = 20) -> Dict:
    """Per-line validation + corpus statistics
    (reference data_processing.py:60-200)."""
    stats = {"lines": 0, "valid": 0, "empty": 0, "errors": [],
             "roles": {}, "total_chars": 0, "max_turns": 0}
    with open(path, encoding="utf-8", errors="replace") as f:
        for i, line in enumerate(f):
            line = line.strip()
            stats["lines"] += 1
            if not line:
                stats["empty"] += 1
                continue
            try:
                row = json.loads(line)
            except json.JSONDecodeError as e:
                if len(stats["errors"]) < max_errors:
                    stats["errors"].append({"line": i + 1, "error": str(e)})
                continue
            msgs = row.get("messages") if isinstance(row, dict) else None
            if not isinstance(msgs, list) or not msgs:
                if isinstance(row, dict) and ("text" in row or "content" in row):
                    stats["valid"] += 1
                    stats["total_chars"] += len(row.get("text") or
                                                row.get("content") or "")
                    continue
                if len(stats["errors"]) < max_errors:
                    stats["errors"].append(
                        {"line": i + 1, "error": "no messages/text field"})
                continue
            ok = True
            for m in msgs:
                if not isinstance(m, dict) or "content" not in m:
                    ok = False
                    break
                role = m.get("role", "user")
                stats["roles"][role] = stats["roles"].get(role, 0) + 1
                stats["total_chars"] += len(str(m["content"]))
            if ok:
                stats["valid"] += 1
                stats["max_turns"] = max(stats["max_turns"], len(msgs))
            elif len(stats["errors"]) < max_errors:
                stats["errors"].append({"line": i + 1, "error": "bad message"})
    stats["ok"] = stats["valid"] > 0 and not stats["errors"]
    return stats


def generate_sample_data(path: str, n: int = 32, seed: int = 0,
                         kind: str = "conversation") -> str:
    """Deterministic synthetic JSONL/text for tests and smoke runs
    (reference data_processing.py:220-273)."""
    rng = random.Random(seed)
    topics = ["the weather", "a recipe", "history", "mathematics", "a poem",
              "debugging", "gardening", "music theory"]
    if kind == "conversation":
        with open(path, "w") as f:
            for i in range(n):
                t = rng.choice(topics)
                row = {"messages": [
                    {"role": "user", "content": f"Tell me about {t} ({i})."},
                    {"role": "assistant",
                     "content": f"Here is what I know about {t}: "
                                + " ".join(rng.choice(topics)
                                           for _ in range(rng.randint(5, 20)))},
                ]}
                f.write(json.dumps(row) + "\n")
    else:
        with open(path, "w") as f:
            for i in range(n):
                f.write(" ".join(rng.choice(topics)
                                 for _ in range(rng.randint(20, 60))) + "\n")
    return path
