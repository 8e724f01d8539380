"""Training reports (reference /root/reference/Src/Main_Scripts/utils/
reporting.py:11-193: data_summary, create_training_report HTML)."""

from __future__ import annotations

import html
import json
import os
import time
from typing import Dict, List, Optional


def data_summary(datasets: Dict[str, object]) -> Dict:
    out = {}
    for name, ds in datasets.items():
        try:
            out[name] = {"examples": len(ds)}
        except TypeError:
            out[name] = {"examples": "streaming"}
    return out


def create_training_report(history: List[Dict], config=None,
                           out_path: str = "training_report.html",
                           extra: Optional[Dict] = None) -> str:
    """Minimal self-contained HTML report: loss table per epoch + config dump
    + final stats. Returns the path written."""
    rows = "".join(
        f"<tr><td>{h.get('epoch')}</td><td>{h.get('mean_loss', float('nan')):.4f}"
        f"</td><td>{h.get('tokens_per_sec', 0):.0f}</td>"
        f"<td>{h.get('duration_s', 0):.1f}</td>"
        f"<td>{h.get('eval', {}).get('loss', '—')}</td></tr>"
        for h in history)
    cfg_json = html.escape(json.dumps(
        config.to_dict() if hasattr(config, "to_dict") else (config or {}),
        indent=2, default=str))
    extra_json = html.escape(json.dumps(extra or {}, indent=2, default=str))
    doc = f"""<!doctype html><html><head><meta charset="utf-8">
<title>LuminaAI-AMD training report</title>
<style>body{{font-family:sans-serif;margin:2em}}table{{border-collapse:collapse}}
td,th{{border:1px solid #999;padding:4px 10px}}pre{{background:#f5f5f5;padding:1em}}
</style></head><body>
<h1>Training report</h1>
<p>generated {time.strftime('%Y-%m-%d %H:%M:%S')}</p>
<h2>Epochs</h2>
<table><tr><th>epoch</th><th>mean loss</th><th>tok/s</th><th>secs</th>
<th>eval loss</th></tr>{rows}</table>
<h2>Extra</h2><pre>{extra_json}</pre>
<h2>Config</h2><pre>{cfg_json}</pre>
</body></html>"""
    os.makedirs(os.path.dirname(out_path) or ".", exist_ok=True)
    with open(out_path, "w") as f:
        f.write(doc)
    return out_path
