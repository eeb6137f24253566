#!/usr/bin/env python3
"""BASELINE config 3: streaming 1 MiB frames between 2 MI355X over xGMI.

Two torchrun ranks; rank 0 streams `--frames` frames of `--frame-mb` MiB
to rank 1. The frame payload lives in GPU memory; the data path is RCCL
point-to-point over xGMI (torch.distributed send/recv, backend nccl);
brpc_amd's streaming layer provides the same credit-window semantics on
the host control path (tests/test_stream.py covers it on CPU).

Launch (2 GPUs):
  python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
      --master-addr 127.0.0.1 tools/stream_xgmi_bench.py --frames 200

On CPU (gloo) the same script measures the host fallback.
Prints one JSON line on rank 0: GB/s over the timed frames.
"""
import argparse
import json
import os
import sys
import time

_local_rank = int(os.environ.get("LOCAL_RANK", "0"))
if int(os.environ.get("WORLD_SIZE", "1")) > 1:
    os.environ.setdefault("HIP_VISIBLE_DEVICES", str(_local_rank))

import torch  # noqa: E402
import torch.distributed as dist  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--frames", type=int, default=100)
    ap.add_argument("--frame-mb", type=int, default=1)
    ap.add_argument("--warmup", type=int, default=10)
    args = ap.parse_args()

    use_gpu = torch.cuda.is_available()
    backend = os.environ.get("BAM_BENCH_BACKEND") or ("nccl" if use_gpu else "gloo")
    use_gpu = use_gpu and backend == "nccl"
    if use_gpu:
        torch.cuda.set_device(0)  # each rank masked to its own GPU
    dist.init_process_group(backend=backend)
    rank = dist.get_rank()
    dev = torch.device("cuda:0") if use_gpu else torch.device("cpu")

    n = args.frame_mb * (1 << 20)
    frame = torch.randint(0, 256, (n,), dtype=torch.uint8, device=dev)
    peer = 1 - rank

    def xfer():
        if rank == 0:
            dist.send(frame, dst=peer)
        else:
            dist.recv(frame, src=peer)

    for _ in range(args.warmup):
        xfer()
    dist.barrier()
    if use_gpu:
        torch.cuda.synchronize()
    t0 = time.monotonic()
    for _ in range(args.frames):
        xfer()
    dist.barrier()
    if use_gpu:
        torch.cuda.synchronize()
    el = time.monotonic() - t0

    if rank == 0:
        gbps = args.frames * n / el / 1e9
        print(json.dumps({
            "metric": "stream_p2p_GBps",
            "value": gbps,
            "frames": args.frames,
            "frame_bytes": n,
            "backend": backend,
            "path": "xGMI (RCCL p2p)" if use_gpu else "host (gloo)",
        }), flush=True)
    dist.destroy_process_group()


if __name__ == "__main__":
    sys.exit(main() or 0)
