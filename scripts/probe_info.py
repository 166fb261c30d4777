"""Which rocSOLVER calls report nonzero info in the real flagship
step, and what are the offending matrices like?"""
import os, sys
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch, torch.nn.functional as F
import torch.distributed as dist
os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
os.environ.setdefault("MASTER_PORT", "29791")
if not dist.is_initialized():
    dist.init_process_group("gloo", world_size=1, rank=0, init_method="env://")
import kfac_pytorch_amd as kfac
from kfac_pytorch_amd.parallel import comm as kcomm
kcomm.init()
from kfac_pytorch_amd.ops import linalg

calls = []
orig = linalg._defer_info
def spy(info):
    calls.append((tuple(info.shape), info.clone()))
    orig(info)
linalg._defer_info = spy

from kfac_pytorch_amd.models.imagenet_resnet import resnet50
model = resnet50().cuda()
pre = kfac.KFAC_EIGEN_DP(model, damping=0.002)
data = torch.randn(32, 3, 224, 224, device="cuda")
tgt = torch.randint(0, 1000, (32,), device="cuda")
for step in range(2):
    model.zero_grad()
    with torch.autocast("cuda", dtype=torch.bfloat16):
        loss = F.cross_entropy(model(data), tgt)
    loss.backward()
    try:
        pre.step()
    except RuntimeError as e:
        print("step", step, "raised:", str(e)[:120])
    torch.cuda.synchronize()
    bad = [(s, v.tolist()) for s, v in calls if int(v.ne(0).sum())]
    print(f"step {step}: {len(calls)} info tensors, bad: {bad[:6]}", flush=True)
    calls.clear()
    linalg._INFO_FLAGS.clear()
