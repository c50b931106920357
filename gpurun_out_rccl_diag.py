import os, glob, sys
sys.path.insert(0, "/root/repo")
os.environ.setdefault("MASTER_ADDR","127.0.0.1"); os.environ.setdefault("MASTER_PORT","29533")
os.environ.setdefault("RANK","0"); os.environ.setdefault("WORLD_SIZE","1")
import torch, torch.distributed as dist
dist.init_process_group("nccl")
torch.cuda.set_device(0)
x = torch.randn(1 << 20, device="cuda")
for _ in range(5): dist.all_reduce(x)
torch.cuda.synchronize(); dist.destroy_process_group()
print("ar-done")
