#!/bin/bash
# Standalone multi-process integration scripts (reference tests/run_tests.sh)
set -e
cd "$(dirname "$0")/.."
TR="python -m torch.distributed.run --nnodes=1 --master-addr 127.0.0.1"

echo "== mnist_dp (2 proc) =="
$TR --nproc-per-node 2 --master-port 29611 tests/standalone/mnist_dp.py
echo "== ta_accelerate (2 proc) =="
$TR --nproc-per-node 2 --master-port 29612 tests/standalone/ta_accelerate.py
echo "== pipeline (4 stages, gc) =="
$TR --nproc-per-node 4 --master-port 29613 tests/standalone/pipeline.py
echo "ALL STANDALONE OK"
