#!/bin/bash
# sbatch wrapper: profile a slurm job step with sofa (reference
# tools/slurmsofa.sh parity).  Usage inside a job script:
#   tools/slurmsofa.sh python train.py --epochs 1
REPO="$(cd "$(dirname "$0")/.." && pwd)"
LOGDIR="${SOFA_LOGDIR:-sofalog-${SLURM_JOB_ID:-local}-$(hostname)}"
exec python3 "$REPO/bin/sofa" stat "$*" --logdir "$LOGDIR" --profile_all_cpus
