#!/usr/bin/env python3
"""Collective bucket-size sweep over RCCL/xGMI (bucket/bench
harness against the 7 x 153 GB/s link model).

Launch one rank per GPU:
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 benchmarks/bench_comm.py
        [--op all_reduce|reduce_scatter|all_gather|all_to_all]
        [--sizes-mb 1,4,16,50,128,512,1024] [--iters 20]

Per size prints achieved algorithm bandwidth (payload GB/s) and bus
bandwidth (ring-equivalent wire GB/s: x2(n-1)/n for all-reduce,
x(n-1)/n for RS/AG). The ZeRO engines default to ~50 MB buckets so
several transfers ride the 7 links concurrently — this harness is how
that number gets re-derived on new topologies. Runs on gloo/CPU for a
smoke check (numbers are meaningless there).
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402
import torch.distributed as dist  # noqa: E402

from luminaai_amd.parallel import comm  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--op", default="all_reduce",
                    choices=["all_reduce", "reduce_scatter", "all_gather",
                             "all_to_all"])
    ap.add_argument("--sizes-mb", default="1,4,16,50,128,512")
    ap.add_argument("--iters", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=5)
    args = ap.parse_args()

    comm.init_distributed()
    world = comm.get_world_size()
    rank = comm.get_rank()
    device = torch.device("cuda", comm.env_local_rank()) \
        if torch.cuda.is_available() else torch.device("cpu")
    if device.type == "cuda":
        torch.cuda.set_device(device)
    dtype = torch.bfloat16 if device.type == "cuda" else torch.float32
    gloo = dist.get_backend() == "gloo"

    def run(op, buf, out):
        if op == "all_reduce":
            dist.all_reduce(buf)
        elif op == "reduce_scatter":
            if gloo:
                dist.all_reduce(buf)        # gloo lacks RS; smoke only
            else:
                dist.reduce_scatter_tensor(out, buf)
        elif op == "all_gather":
            if gloo:
                dist.all_gather(list(buf.chunk(world)), out)
            else:
                dist.all_gather_into_tensor(buf, out)
        elif op == "all_to_all":
            dist.all_to_all_single(out, buf)

    results = []
    for mb in [float(s) for s in args.sizes_mb.split(",")]:
        n = int(mb * 1e6 / buf_elem_size(dtype))
        n = (n // (world * 256)) * (world * 256) or world * 256
        buf = torch.randn(n, device=device).to(dtype)
        out = torch.empty(n // world, device=device, dtype=dtype) \
            if args.op in ("reduce_scatter",) else torch.empty_like(buf)
        if args.op == "all_gather":
            out = torch.randn(n // world, device=device).to(dtype)
        for _ in range(args.warmup):
            run(args.op, buf, out)
        if device.type == "cuda":
            torch.cuda.synchronize()
        dist.barrier()
        t0 = time.perf_counter()
        for _ in range(args.iters):
            run(args.op, buf, out)
        if device.type == "cuda":
            torch.cuda.synchronize()
        dist.barrier()
        dt = (time.perf_counter() - t0) / args.iters
        payload = n * buf.element_size()
        alg = payload / dt / 1e9
        factor = {"all_reduce": 2 * (world - 1) / world,
                  "reduce_scatter": (world - 1) / world,
                  "all_gather": (world - 1) / world,
                  "all_to_all": (world - 1) / world}[args.op]
        results.append({"size_mb": round(payload / 1e6, 1),
                        "ms": round(dt * 1e3, 3),
                        "alg_GBps": round(alg, 1),
                        "bus_GBps": round(alg * factor, 1)})
    if rank == 0:
        print(json.dumps({"op": args.op, "world": world,
                          "backend": dist.get_backend(),
                          "device": str(device), "sweep": results}))
    comm.cleanup()


def buf_elem_size(dtype):
    return torch.empty(0, dtype=dtype).element_size()


if __name__ == "__main__":
    main()
