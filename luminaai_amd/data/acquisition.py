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
# Per-source cleaning pipelines (reference multi_source_dataset.py:277-1350:
# WikipediaProcessor :277, GutenbergProcessor :511, ArXivProcessor :616,
# StackOverflowProcessor :729, PubMedProcessor :852, OpenWebTextProcessor
# :1012, PhilPapersProcessor :1125, CommonCrawlNewsProcessor :1229).  The
# download legs are network-gated; these run on local dumps.
_WS = re.compile(r"[ \t]+")
_MULTI_NL = re.compile(r"\n{3,}")


def _clean_generic(text: str) -> str:
    text = _WS.sub(" ", text)
    text = _MULTI_NL.sub("\n\n", text)
    return text.strip()


def _strip_html(text: str) -> str:
    text = re.sub(r"<code>.*?</code>", "[CODE]", text, flags=re.S | re.I)
    text = re.sub(r"<[^>]+>", "", text)
    for ent, ch in (("&lt;", "<"), ("&gt;", ">"), ("&amp;", "&"),
                    ("&quot;", '"'), ("&#39;", "'"), ("&nbsp;", " ")):
        text = text.replace(ent, ch)
    return text


def _clean_wikipedia(text: str) -> str:
    """Wikitext -> plain prose (reference clean_wiki_text :316)."""
    text = re.sub(r"<!--.*?-->", "", text, flags=re.S)
    text = re.sub(r"<ref[^>]*>.*?</ref>", "", text, flags=re.S | re.I)
    text = re.sub(r"<ref[^/>]*/>", "", text, flags=re.I)
    for _ in range(5):                       # nested templates
        text = re.sub(r"\{\{[^{}]*\}\}", "", text)
    text = re.sub(r"\{\{.*?\}\}", "", text, flags=re.S)
    text = re.sub(r"\{\|.*?\|\}", "", text, flags=re.S)    # tables/infoboxes
    text = re.sub(r"\[\[(?:File|Image|\u0424\u0430\u0439\u043b):.*?\]\]", "", text,
                  flags=re.I | re.S)
    text = re.sub(r"<gallery.*?>.*?</gallery>", "", text, flags=re.S | re.I)
    text = re.sub(r"\[\[Category:.*?\]\]", "", text, flags=re.I)
    text = re.sub(r"\[\[(?:[^|\]]*\|)?([^\]]+)\]\]", r"\1", text)  # links
    text = re.sub(r"\[http[^\]]*\]", "", text)
    text = re.sub(r"http[s]?://\S+", "", text)
    text = re.sub(r"<[^>]+>", "", text)
    # drop trailing reference sections, flatten remaining headers
    text = re.sub(r"==+ *(References|External links|See also|Further "
                  r"reading|Notes|Bibliography) *==+.*", "", text, flags=re.S)
    text = re.sub(r"==+ *([^=\n]+?) *==+", r"\1.", text)
    text = text.replace("]]", "").replace("[[", "")
    text = re.sub(r"\'{2,}", "", text)       # bold/italic quotes
    return _clean_generic(text)


def _clean_gutenberg(text: str) -> str:
    """Strip Project Gutenberg boilerplate (reference :552)."""
    m = re.search(r"\*\*\* ?START OF (THIS|THE) PROJECT GUTENBERG EBOOK"
                  r".*?\*\*\*", text, re.I | re.S)
    if m:
        text = text[m.end():]
    m = re.search(r"\*\*\* ?END OF (THIS|THE) PROJECT GUTENBERG EBOOK",
                  text, re.I)
    if m:
        text = text[:m.start()]
    text = re.sub(r"\[Illustration:?[^\]]*\]", "", text)
    return _clean_generic(text)


_LATEX_CMD = re.compile(r"\\[a-zA-Z]+\*?(\[[^\]]*\])?(\{[^{}]*\})?")


def _clean_arxiv(row) -> str:
    """Title + abstract with LaTeX de-noising (reference ArXivProcessor
    :616 -- abstracts only, 'Title: .. Abstract: ..' records)."""
    if isinstance(row, dict):
        title = (row.get("title") or "").strip()
        abstract = (row.get("abstract") or row.get("summary") or "").strip()
    else:
        title, abstract = "", str(row)
    body = f"Title: {title}\n\nAbstract: {abstract}" if title else abstract
    body = re.sub(r"\$+[^$]*\$+", " [MATH] ", body)
    body = _LATEX_CMD.sub(" ", body)
    body = body.replace("{", "").replace("}", "")
    return _clean_generic(body)


def _clean_stackoverflow(row) -> str:
    """Q/A records -> 'Question: .. Answer: ..' with HTML stripped and
    code spans tokenised (reference StackOverflowProcessor :729)."""
    if isinstance(row, dict):
        q = row.get("question") or row.get("title") or ""
        body = row.get("body") or ""
        a = row.get("answer") or row.get("accepted_answer") or ""
        parts = []
        if q:
            parts.append("Question: " + _strip_html(q))
        if body:
            parts.append(_strip_html(body))
        if a:
            parts.append("Answer: " + _strip_html(a))
        return _clean_generic("\n".join(parts))
    return _clean_generic(_strip_html(str(row)))


_PUBMED_LABELS = re.compile(
    r"\b(BACKGROUND|OBJECTIVES?|METHODS?|RESULTS?|CONCLUSIONS?|PURPOSE|"
    r"DESIGN|SETTING|PARTICIPANTS|INTERVENTIONS?|MAIN OUTCOME MEASURES?|"
    r"MEASUREMENTS)\s*:\s*")


def _clean_pubmed(row) -> str:
    """Structured-abstract labels stripped, title+abstract records
    (reference PubMedProcessor :852)."""
    if isinstance(row, dict):
        title = (row.get("title") or "").strip().rstrip(".")
        abstract = (row.get("abstract") or row.get("text") or "").strip()
        body = f"{title}. {abstract}" if title else abstract
    else:
        body = str(row)
    body = _PUBMED_LABELS.sub("", body)
    body = re.sub(r"\[[0-9,\- ]+\]", "", body)      # citation brackets
    return _clean_generic(body)


_NAVLINE = re.compile(
    r"^(\s*(home|menu|login|sign ?in|sign ?up|subscribe|share|tweet|"
    r"advertisement|cookie[s]? (policy|notice)|privacy policy|terms of "
    r"(use|service)|all rights reserved).*|[\W\d\s]*)$", re.I)


def _clean_openwebtext(text: str) -> str:
    """Boilerplate/navigation line filtering + dedupe (reference
    OpenWebTextProcessor :1012)."""
    seen = set()
    keep = []
    for line in _strip_html(text).splitlines():
        ls = line.strip()
        if len(ls) < 3 or _NAVLINE.match(ls):
            continue
        # drop exact repeats (nav fragments recur on every page)
        if len(ls) < 80:
            if ls in seen:
                continue
            seen.add(ls)
        keep.append(ls)
    return _clean_generic("\n".join(keep))


def _clean_philpapers(row) -> str:
    """Philosophy abstracts: same record shape as arXiv (reference
    PhilPapersProcessor :1125)."""
    return _clean_arxiv(row)


_BYLINE = re.compile(r"^(by [A-Z][\w.\- ]+|published:? .*|updated:? .*|"
                     r"\d{1,2} [A-Z][a-z]+ \d{4}.*)$", re.I | re.M)


def _clean_cc_news(row) -> str:
    """Headline + body with bylines/dates and site boilerplate removed
    (reference CommonCrawlNewsProcessor :1229)."""
    if isinstance(row, dict):
        title = (row.get("title") or "").strip()
        body = (row.get("text") or row.get("body") or "").strip()
        text = f"{title}\n\n{body}" if title else body
    else:
        text = str(row)
    text = _BYLINE.sub("", _strip_html(text))
    return _clean_openwebtext(text)


def _clean_code_qa(text: str) -> str:    # retained alias (round-1 API)
    return _clean_generic(_strip_html(text))


SOURCE_PROCESSORS: Dict[str, Callable] = {
    "wikipedia": _clean_wikipedia,
    "gutenberg": _clean_gutenberg,
    "arxiv": _clean_arxiv,
    "stackoverflow": _clean_stackoverflow,
    "pubmed": _clean_pubmed,
    "openwebtext": _clean_openwebtext,
    "philpapers": _clean_philpapers,
    "cc_news": _clean_cc_news,
}

# processors that want the parsed JSON row rather than a text line
_ROW_PROCESSORS = {"arxiv", "stackoverflow", "pubmed", "philpapers",
                   "cc_news"}


def prepare_text_corpus(input_path: str, out_path: str, source: str = "generic",
                        min_chars: int = 200) -> Dict:
    """Clean a local raw-text/JSONL dump into a training .txt corpus
    (reference multi_source_dataset.py per-source processors)."""
    proc = SOURCE_PROCESSORS.get(source, _clean_generic)
    row_mode = source in _ROW_PROCESSORS
    n_in = n_out = 0
    os.makedirs(os.path.dirname(out_path) or ".", exist_ok=True)
    with open(input_path, encoding="utf-8", errors="replace") as fin, \
            open(out_path, "w") as fout:
        for line in fin:
            n_in += 1
            payload = line
            if line.lstrip().startswith("{"):
                try:
                    row = json.loads(line)
                    payload = row if row_mode else (
                        row.get("text") or row.get("content") or "")
                except json.JSONDecodeError:
                    pass
            text = proc(payload)
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
