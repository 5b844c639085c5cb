#!/usr/bin/env python3
"""DDP training example with the cgx compressed-allreduce backend.

Parity with the reference examples/cifar_train.py, minus torchvision/datasets
(this image has no network): trains ResNet-18 on synthetic CIFAR-shaped data.

Launch:  torchrun --standalone --nproc-per-node 8 examples/train_ddp.py \
             --quantization-bits 4 --quantization-bucket-size 1024
"""

import argparse
import os

os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist
import torch.nn as nn

import torch_cgx_amd
from torch_cgx_amd.models import resnet18


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--epochs", type=int, default=2)
    p.add_argument("--steps-per-epoch", type=int, default=50)
    p.add_argument("--batch-size", type=int, default=128)
    p.add_argument("--lr", type=float, default=0.1)
    p.add_argument("--quantization-bits", type=int, default=4)
    p.add_argument("--quantization-bucket-size", type=int, default=1024)
    p.add_argument("--dist-backend", default="cgx")
    p.add_argument("--error-feedback", action="store_true",
                   help="fold the quantization residual into the next step")
    p.add_argument("--adaptive", action="store_true",
                   help="step-scheduled bits: fp32 warmup -> 8 -> target")
    args = p.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    torch.cuda.set_device(local_rank)
    dist.init_process_group(args.dist_backend, rank=rank, world_size=world)

    model = resnet18(num_classes=100).cuda()
    model = nn.parallel.DistributedDataParallel(model,
                                                device_ids=[local_rank])
    if args.adaptive:
        state = torch_cgx_amd.AdaptiveCGXState(
            None, schedule=[(0, 32), (10, 8), (30, args.quantization_bits)],
            bucket_size=args.quantization_bucket_size,
            error_feedback=args.error_feedback)
        model.register_comm_hook(state, torch_cgx_amd.adaptive_cgx_hook)
    else:
        params = {"bits": args.quantization_bits,
                  "bucket_size": args.quantization_bucket_size}
        if args.error_feedback:
            params["error_feedback"] = True
        state = torch_cgx_amd.CGXState(None, layer_min_size=1024,
                                       compression_params=params)
        model.register_comm_hook(state, torch_cgx_amd.cgx_hook)

    opt = torch.optim.SGD(model.parameters(), lr=args.lr, momentum=0.9,
                          weight_decay=5e-4)
    torch.manual_seed(1 + rank)

    for epoch in range(args.epochs):
        t0 = time.perf_counter()
        total, correct, loss_sum = 0, 0, 0.0
        for _ in range(args.steps_per_epoch):
            x = torch.randn(args.batch_size, 3, 32, 32, device="cuda")
            y = torch.randint(0, 100, (args.batch_size,), device="cuda")
            with torch.autocast("cuda", dtype=torch.bfloat16):
                logits = model(x)
                loss = nn.functional.cross_entropy(logits, y)
            opt.zero_grad(set_to_none=True)
            loss.backward()
            opt.step()
            loss_sum += loss.item()
            correct += (logits.argmax(1) == y).sum().item()
            total += y.numel()
        torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        if rank == 0:
            ips = args.batch_size * args.steps_per_epoch * world / dt
            print(f"epoch {epoch}: loss={loss_sum/args.steps_per_epoch:.3f} "
                  f"acc={correct/total:.3f} {ips:.0f} img/s", flush=True)

    dist.destroy_process_group()


if __name__ == "__main__":
    main()
