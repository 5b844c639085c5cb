#!/bin/bash
# DDP training example with compressed gradients (parity with the
# reference's run_cifar.sh, torchrun instead of mpirun).
NGPU=${1:-8}
torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node "$NGPU" \
    examples/train_ddp.py --quantization-bits 4 --quantization-bucket-size 1024
