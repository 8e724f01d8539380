#!/usr/bin/env python3
"""Inference server — `python serve.py --checkpoint PATH [--port 8000]`.
OpenAI-compatible /v1/completions and /v1/chat/completions with SSE
streaming over the KV-cached (hipGraph on GPU) decode engine."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from luminaai_amd.inference.server import main  # noqa: E402

if __name__ == "__main__":
    main()
