#!/usr/bin/env bash
# Launch the DDP example on N GPUs (parity with the reference's run_cifar.sh,
# torchrun instead of mpirun — no MPI on this stack).
N=${1:-8}
shift || true
torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node "$N" \
    examples/train_ddp.py --quantization-bits 4 --quantization-bucket-size 1024 "$@"
