#!/bin/bash
# Example: profile a (reduced-layer) Llama-architecture training step with
# the full surface: AISI, call stacks, KFD events (BASELINE config 5 shape).
set -e
cd "$(dirname "$0")/.."
python3 bin/sofa stat "python3 -m sofa_amd.workloads.llama --layers 8 --steps 4 --batch 2 --seq 2048" \
    --logdir sofalog-llama --enable_aisi --num_iterations 4 --call_stacks --enable_kfd_trace
echo "now: python3 bin/sofa viz --logdir sofalog-llama"
