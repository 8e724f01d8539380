#!/usr/bin/env python3
"""Chat REPL — `python chat.py [--checkpoint PATH]` (reference Chat.py).
Loads the newest checkpoint by default, infers the architecture from tensor
shapes, and decodes incrementally with per-layer KV caches."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from luminaai_amd.inference.chat import main  # noqa: E402

if __name__ == "__main__":
    main()
