#!/bin/bash
# One process per MI355X over RCCL/xGMI (parity with the reference's
# start_distributed.sh, scaled to the 8-GPU node).
NPROC=${NPROC:-8}
OMP_NUM_THREADS=1 nohup python -m torch.distributed.run \
  --nnodes 1 --nproc_per_node "$NPROC" --master-addr 127.0.0.1 \
  main.py --mode train_test "$@" > nohup.out 2>&1 &
echo "pid: $!"
