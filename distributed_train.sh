#!/bin/bash
# Launch train.py with one process per GPU over RCCL/xGMI.
# Usage: ./distributed_train.sh <num-gpus> [train.py args...]
NUM_PROC=$1
shift
torchrun --standalone --nproc_per_node=$NUM_PROC --local-addr 127.0.0.1 train.py "$@"
