#!/bin/bash
# Reproduce the BASELINE.json measurement matrix on an 8x MI355X node.
# Usage: bash benchmarks/run_scale.sh
set -x
for N in 1 2 4 8; do
  python -m torch.distributed.run --nnodes=1 --nproc-per-node $N \
      --master-addr 127.0.0.1 --master-port 29531 \
      bench.py --gpus $N --steps 20 --warmup 5                      # config 3
done
for N in 2 8; do
  # config 2 (64M fp16 4-bit bucket 512) and config 5 (256M sweep, fusion 64MB)
  python -m torch.distributed.run --nnodes=1 --nproc-per-node $N \
      --master-addr 127.0.0.1 --master-port 29532 \
      benchmarks/allreduce_bench.py --size 67108864 --dtype float16 \
      --bits 4 --bucket-size 512
  python -m torch.distributed.run --nnodes=1 --nproc-per-node $N \
      --master-addr 127.0.0.1 --master-port 29533 \
      benchmarks/allreduce_bench.py --size 268435456 --dtype float16 \
      --bits 1 2 4 8 32 --fusion-mb 64
  # fp32 RCCL baseline on identical shapes
  python -m torch.distributed.run --nnodes=1 --nproc-per-node $N \
      --master-addr 127.0.0.1 --master-port 29534 \
      benchmarks/allreduce_bench.py --size 268435456 --dtype float16 \
      --bits 32 --backend nccl
done
python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
    --master-addr 127.0.0.1 --master-port 29535 \
    bench.py --gpus 8 --model bert-large --bits 8 --show-registry \
    --steps 20 --warmup 5                                            # config 4
