#!/bin/bash
# Example: profile a ResNet-50 training burst on one MI355X and open the
# dashboard (BASELINE config 3).
set -e
cd "$(dirname "$0")/.."
python bin/sofa stat "python -c '
import sys, torch; sys.path.insert(0, \".\")
from sofa_amd.workloads.resnet import build_resnet50
m = build_resnet50(); opt = torch.optim.SGD(m.parameters(), lr=0.1)
x = torch.randn(64, 3, 224, 224, device=\"cuda\").to(memory_format=torch.channels_last)
t = torch.randint(0, 1000, (64,), device=\"cuda\")
for _ in range(10):
    with torch.autocast(\"cuda\", dtype=torch.bfloat16):
        loss = torch.nn.functional.cross_entropy(m(x), t)
    opt.zero_grad(); loss.backward(); opt.step()
torch.cuda.synchronize()
'" --logdir sofalog-resnet --enable_aisi --num_iterations 10
echo "now: python bin/sofa viz --logdir sofalog-resnet"
