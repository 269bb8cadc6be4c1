#!/bin/bash
# Reproduce the numbers in profiles/RESULTS_r02.md (and _r01) on an MI355X box.
# Usage: bash scripts/run_benchmarks.sh [quick]
set -x
QUICK=${1:-}
ITERS=300; STEPS=60
if [ "$QUICK" = "quick" ]; then ITERS=50; STEPS=20; fi

# headline: CG on 5-pt Poisson (BASELINE.json config)
python bench.py --steps $STEPS --warmup 10 --nx 16384

# reference's own headline config (75.9 it/s on V100)
python examples/pde.py -nx 6000 -ny 6000 -throughput -max_iter $ITERS

# SpMV microbenchmark (347.7 it/s on V100)
python examples/dot_microbenchmark.py -n 10000000 -iters $ITERS

# GMG (37.2 it/s at N=4500 on V100)
python examples/gmg.py -N 4095 -maxiter 100
python examples/gmg.py -N 511 -dim 3 -maxiter 60

# AMG + Galerkin SpGEMM
python examples/amg.py -n 1048576 -maxiter 200
python examples/spgemm_microbenchmark.py -nx 2047 -iters 10

# quantum MIS demo (6x6 = FULL 5.6M-state Hilbert space; 9x9 truncated)
python examples/quantum_mis.py -l 6 -T 4.0
python examples/quantum_mis.py -l 9 -T 4.0 -kmax 4

# multi-vector SpMM: auto BSR-MFMA route vs lane-tiled (profiles/MFMA_r02.md)
python examples/dot_microbenchmark.py -op spmm -k 32 -n 4000000 -iters 50 -warmup 10
SPARSE_NO_BSR=1 python examples/dot_microbenchmark.py -op spmm -k 32 -n 4000000 -iters 50 -warmup 10

# MFMA kernel A/B + lane-map probe (standalone)
hipcc --offload-arch=gfx950 -O3 tools/mfma_spmm_bench.hip -o /tmp/mfma_bench
/tmp/mfma_bench --probe && /tmp/mfma_bench 1048576 30

# capacity: 604M rows / 3.02B nnz on ONE GPU (288 GB sizing)
python bench.py --nx 24576 --steps 30 --warmup 5

# fp32 option
python bench.py --nx 8192 --dtype fp32 --steps $STEPS --warmup 10

# 8-GPU strong scaling (run on an 8-GPU node; the driver does this too)
# python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
#     --master-addr 127.0.0.1 bench.py --gpus 8 --steps $STEPS --warmup 15
