#!/bin/bash
# Snapshot a known-good recording into demo/ (reference tools/build_demo.sh).
set -e
cd "$(dirname "$0")/.."
python3 bin/sofa stat "dd if=/dev/zero of=/tmp/sofa_demo.out bs=50M count=20" \
    --logdir /tmp/sofa_demo_log --profile_all_cpus
rm -rf demo/logdir
mkdir -p demo
cp -r /tmp/sofa_demo_log demo/logdir
rm -f demo/logdir/kallsyms demo/logdir/cpusamples.scs
echo "demo at demo/logdir — view with: python3 bin/sofa viz --logdir demo/sofalog"
