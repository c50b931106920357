#!/bin/bash
# Diagnose the lite-collector edge cases on a real MI355X box:
#   A. hsalite alone — wrapper counters (which copy entry points fire?)
#   B. hsalite + SDK(rccl/copy-only) — tool-chaining: do kernels still land?
#   C. sofa stat --gpu_tracer lite — full pipeline stdout (round-2 test failed
#      with no gputrace.csv; capture the preprocess warnings)
#   D. SOFA_LITE_REPLACE_SIGNALS=1 — close the ~0.6% signal-carrying kernel
#      gap; bounded by timeout in case forwarding stalls a sync point
set -x
cd /root/repo
OUT=gpurun_out/lite_probe
mkdir -p $OUT
LITE=$PWD/sofa_amd/native/lib/libsofahsalite.so
SDK=$PWD/sofa_amd/native/lib/libsofatracer.so

SNIP='import torch
x = torch.randn(1024, 1024, device="cuda")
for _ in range(10):
    x = x @ x; x = x / x.norm()
h = x.cpu()
torch.cuda.synchronize()
print("okay", float(h.sum()))'

echo "=== A: hsalite alone ==="
rm -rf $OUT/a; mkdir -p $OUT/a
SOFA_LOGDIR=$OUT/a SOFA_LITE_DEBUG=1 HSA_TOOLS_LIB=$LITE \
  timeout 300 python -c "$SNIP" > $OUT/a/stdout.log 2> $OUT/a/stderr.log
tail -3 $OUT/a/stderr.log

echo "=== B: hsalite + SDK copy/rccl-only ==="
rm -rf $OUT/b; mkdir -p $OUT/b
SOFA_LOGDIR=$OUT/b SOFA_LITE_DEBUG=1 HSA_TOOLS_LIB=$LITE \
  ROCP_TOOL_LIBRARIES=$SDK SOFA_TRACE_DISPATCH=0 SOFA_TRACE_COPY=1 SOFA_TRACE_RCCL=1 \
  timeout 300 python -c "$SNIP" > $OUT/b/stdout.log 2> $OUT/b/stderr.log
tail -3 $OUT/b/stderr.log

echo "=== C: sofa stat --gpu_tracer lite ==="
rm -rf $OUT/c
timeout 600 python bin/sofa stat "python -c \"import torch; x=torch.randn(512,512,device='cuda'); [x.matmul(x) for _ in range(20)]; torch.cuda.synchronize(); print('okay')\"" \
  --logdir $OUT/c --gpu_tracer lite --verbose > $OUT/c_stdout.log 2> $OUT/c_stderr.log
tail -5 $OUT/c_stdout.log
ls -la $OUT/c/ | head -30

echo "=== D: replace-signals mode ==="
rm -rf $OUT/d; mkdir -p $OUT/d
SOFA_LOGDIR=$OUT/d SOFA_LITE_DEBUG=1 SOFA_LITE_REPLACE_SIGNALS=1 HSA_TOOLS_LIB=$LITE \
  timeout 300 python -c "$SNIP" > $OUT/d/stdout.log 2> $OUT/d/stderr.log
tail -3 $OUT/d/stderr.log

echo "=== parse summaries ==="
python - <<'EOF'
import glob, sys
sys.path.insert(0, '.')
from sofa_amd.preprocess.sgt import parse_sgt
for tag in 'abd':
    for f in glob.glob(f'gpurun_out/lite_probe/{tag}/gputrace_*.sgt'):
        s = parse_sgt(f)
        kind = 'lite' if '_lite' in f else 'sdk'
        print(tag, kind, 'kernels', len(s.kernels), 'copies', len(s.copies))
EOF
