#!/bin/bash
# Legacy multi-GPU launcher kept for CLI compatibility; forwards to the
# task-based train.py (see distributed_train.sh).
NUM_PROC=$1
shift
torchrun --standalone --nproc_per_node=$NUM_PROC --local-addr 127.0.0.1 legacy_train.py "$@"
