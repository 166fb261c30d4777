"""Did the pred hipGraph capture? And what is pred actually bound by?"""
import os, sys, time
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch, torch.distributed as dist, torch.nn.functional as F
os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
os.environ.setdefault("MASTER_PORT", "29764")
if not dist.is_initialized():
    dist.init_process_group("gloo", world_size=1, rank=0, init_method="env://")
import kfac_pytorch_amd as kfac
import kfac_pytorch_amd.backend as backend
backend.init("Torch")
from kfac_pytorch_amd.models.imagenet_resnet import resnet50
model = resnet50().cuda()
pre = kfac.KFAC_EIGEN_DP(model, damping=0.002)
x = torch.randn(32, 3, 224, 224, device="cuda")
y = torch.randint(0, 1000, (32,), device="cuda")
for step in range(3):
    model.zero_grad(set_to_none=False)
    with torch.autocast("cuda", dtype=torch.bfloat16):
        F.cross_entropy(model(x), y).backward()
    pre.step()
torch.cuda.synchronize()
print("graph_disabled:", getattr(pre, "_graph_disabled", False))
print("graphs:", {k: (v is not None) for k, v in
                  getattr(pre, "_graphs", {}).items()})
# time pred phase alone: replay vs eager
for flag in ("1", "0"):
    os.environ["KFAC_PRED_GRAPH"] = flag
    pre._graphs = {}
    pre._graph_disabled = False
    pre._compute_pred(); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(5):
        pre._compute_pred()
    torch.cuda.synchronize()
    print(f"pred KFAC_PRED_GRAPH={flag}: "
          f"{(time.perf_counter()-t0)/5*1e3:.2f} ms", flush=True)
