#!/bin/bash
# Snapshot a known-good recording into demo/ (reference tools/build_demo.sh).
set -e
cd "$(dirname "$0")/.."
python3 bin/sofa stat "dd if=/dev/zero of=/tmp/sofa_demo.out bs=50M count=20" \
    --logdir /tmp/sofa_demo_log --profile_all_cpus
rm -rf demo/sofalog
mkdir -p demo
cp -r /tmp/sofa_demo_log demo/sofalog
echo "demo at demo/sofalog — view with: python3 bin/sofa viz --logdir demo/sofalog"
