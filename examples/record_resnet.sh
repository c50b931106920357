#!/bin/bash
# Example: profile a ResNet-50 training burst on one MI355X and open the
# dashboard (BASELINE config 3).
set -e
cd "$(dirname "$0")/.."
python3 bin/sofa stat "python3 -m sofa_amd.workloads.resnet_burst --iters 10" \
    --logdir sofalog-resnet --enable_aisi --num_iterations 10 --call_stacks
echo "now: python3 bin/sofa viz --logdir sofalog-resnet"
