#!/bin/bash
# Round-2 GPU probe #3:
#  A. copy fix: hsalite must now record the D2H (null-signal on_engine path)
#  B. hybrid-load diagnosis: why does ROCP_TOOL_LIBRARIES block hsalite's
#     OnLoad?  (constructor + OnLoad prints + ROCr load-failure reporting)
#  C. lite-mode DDP world-1: RCCL debug-log channel end-to-end through
#     sofa stat (rccltrace.csv from NCCL_DEBUG=INFO COLL)
set -x
cd /root/repo
OUT=gpurun_out/lite_probe2
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

echo "=== A: copy fix ==="
rm -rf $OUT/a; mkdir -p $OUT/a
SOFA_LOGDIR=$OUT/a SOFA_LITE_DEBUG=1 HSA_TOOLS_LIB=$LITE \
  timeout 300 python -c "$SNIP" > $OUT/a/stdout.log 2> $OUT/a/stderr.log
grep sofahsalite $OUT/a/stderr.log | tail -3

echo "=== B: hybrid load diagnosis ==="
rm -rf $OUT/b; mkdir -p $OUT/b
SOFA_LOGDIR=$OUT/b SOFA_LITE_DEBUG=1 HSA_TOOLS_LIB=$LITE \
  ROCP_TOOL_LIBRARIES=$SDK SOFA_TRACE_DISPATCH=0 \
  HSA_TOOLS_REPORT_LOAD_FAILURE=1 \
  timeout 300 python -c "import os, torch; x = torch.randn(64, device='cuda'); torch.cuda.synchronize(); print('HSA_TOOLS_LIB=', os.environ.get('HSA_TOOLS_LIB'))" \
  > $OUT/b/stdout.log 2> $OUT/b/stderr.log
cat $OUT/b/stdout.log
grep -iE "sofahsalite|failed to load|Tool lib" $OUT/b/stderr.log | head -5

echo "=== C: lite-mode RCCL log channel via sofa stat ==="
rm -rf $OUT/c
cat > /tmp/ar.py <<'EOF'
import os, torch, torch.distributed as dist
os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
os.environ.setdefault("MASTER_PORT", "29571")
os.environ.setdefault("RANK", "0")
os.environ.setdefault("WORLD_SIZE", "1")
dist.init_process_group("nccl")
torch.cuda.set_device(0)
x = torch.randn(1 << 20, device="cuda")
for _ in range(5):
    dist.all_reduce(x)
torch.cuda.synchronize()
dist.destroy_process_group()
print("ar-done")
EOF
timeout 600 python bin/sofa stat "python /tmp/ar.py" --logdir $OUT/c --gpu_tracer lite > $OUT/c_stdout.log 2> $OUT/c_stderr.log
grep -cE "Complete" $OUT/c_stdout.log
ls $OUT/c/ | grep -E "rccl|sgt"
head -3 $OUT/c/rccltrace.csv 2>/dev/null

echo "=== parse ==="
python - <<'EOF'
import glob, sys
sys.path.insert(0, '.')
from sofa_amd.preprocess.sgt import parse_sgt
for f in glob.glob('gpurun_out/lite_probe2/a/gputrace_*_lite.sgt'):
    s = parse_sgt(f)
    print('A lite kernels', len(s.kernels), 'copies', len(s.copies))
    if len(s.copies):
        c = s.copies[0]
        print('  copy op', c['op'], 'bytes', c['bytes'], 'dur_us',
              (c['end_ns']-c['start_ns'])/1e3)
import pandas as pd, os
p = 'gpurun_out/lite_probe2/c/rccltrace.csv'
if os.path.isfile(p):
    df = pd.read_csv(p)
    print('C rccltrace rows', len(df))
    print(df[['name','payload','deviceId']].head(3).to_string())
p = 'gpurun_out/lite_probe2/c/gputrace.csv'
if os.path.isfile(p):
    df = pd.read_csv(p)
    print('C gputrace kernels', (df['copyKind']==0).sum())
    print('C rccl kernels', df['name'].str.contains('ncclDevKernel').sum())
EOF
