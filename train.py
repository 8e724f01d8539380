#!/usr/bin/env python3
"""Training CLI — `python train.py --preset b1 --train-data data/train.jsonl`.
Multi-GPU: `python -m torch.distributed.run --nnodes=1 --nproc-per-node 8
--master-addr 127.0.0.1 train.py --preset b1_moe ...` (one rank per MI355X
over RCCL). See luminaai_amd/main.py."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from luminaai_amd.main import main  # noqa: E402

if __name__ == "__main__":
    main()
