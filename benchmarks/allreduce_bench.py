#!/usr/bin/env python3
"""Allreduce micro-benchmark.

Measures the cgx compressed allreduce on a flat gradient buffer across a bits
sweep, reporting per-bits latency and effective bandwidths:

  algbw = payload_bytes / time
  busbw = algbw * 2 * (ws - 1) / ws      (ring-equivalent wire bandwidth)

Launch (driver or user):
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 benchmarks/allreduce_bench.py \
      --size 268435456 --dtype float16 --bits 1 2 4 8 32

--backend nccl gives the uncompressed RCCL baseline on identical shapes.
With WORLD_SIZE==1 it times the quantize/dequantize kernels only.
"""

import argparse
import json
import os

os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist

DT = {"float32": torch.float32, "float16": torch.float16,
      "bfloat16": torch.bfloat16}


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--size", type=int, default=64 * 1024 * 1024)
    p.add_argument("--dtype", default="float16", choices=DT)
    p.add_argument("--bits", type=int, nargs="+", default=[1, 2, 4, 8, 32])
    p.add_argument("--bucket-size", type=int, default=512)
    p.add_argument("--fusion-mb", type=int, default=64)
    p.add_argument("--reps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--backend", default="cgx", choices=["cgx", "nccl"])
    args = p.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    assert torch.cuda.is_available()
    torch.cuda.set_device(local_rank)
    device = torch.device("cuda", local_rank)
    dtype = DT[args.dtype]
    es = torch.tensor([], dtype=dtype).element_size()
    os.environ["CGX_FUSION_BUFFER_SIZE_MB"] = str(args.fusion_mb)
    os.environ["CGX_COMPRESSION_BUCKET_SIZE"] = str(args.bucket_size)

    if world > 1:
        if args.backend == "cgx":
            import torch_cgx_amd  # noqa: F401
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        dist.init_process_group(args.backend, rank=rank, world_size=world)

    torch.manual_seed(42 + rank)
    base = torch.randn(args.size, dtype=torch.float32).to(dtype).to(device)

    results = []
    for bits in args.bits:
        os.environ["CGX_COMPRESSION_QUANTIZATION_BITS"] = str(bits)
        if world > 1:
            def run_once():
                t = base.clone()
                dist.all_reduce(t)
                return t
            for _ in range(args.warmup):
                run_once()
            dist.barrier()
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(args.reps):
                run_once()
            dist.barrier()
            torch.cuda.synchronize()
            dt = (time.perf_counter() - t0) / args.reps
            e = torch.tensor([dt], device=device)
            dist.all_reduce(e, op=dist.ReduceOp.MAX)
            dt = e.item()
        else:
            # kernel-only timing at ws=1
            from torch_cgx_amd import _C
            if bits > 8:
                continue
            comp = _C.quantize(base, bits, args.bucket_size, True, 0)
            out = torch.empty_like(base)
            for _ in range(args.warmup):
                comp = _C.quantize(base, bits, args.bucket_size, True, 0)
                _C.dequantize(comp, out, bits, args.bucket_size, False)
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(args.reps):
                comp = _C.quantize(base, bits, args.bucket_size, True, 0)
                _C.dequantize(comp, out, bits, args.bucket_size, False)
            torch.cuda.synchronize()
            dt = (time.perf_counter() - t0) / args.reps

        payload = args.size * es
        algbw = payload / dt / 1e9
        busbw = algbw * 2 * (world - 1) / world if world > 1 else algbw
        if rank == 0:
            r = {"bench": "allreduce", "backend": args.backend,
                 "bits": bits, "size_elems": args.size, "dtype": args.dtype,
                 "bucket_size": args.bucket_size, "n_gpus": world,
                 "ms": round(dt * 1e3, 3), "algbw_GBs": round(algbw, 1),
                 "busbw_GBs": round(busbw, 1)}
            results.append(r)
            print(json.dumps(r), flush=True)

    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
