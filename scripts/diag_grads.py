"""Diagnostic: per-layer grad error of (ours bf16) and (torch autocast
bf16 twin) against the fp32 twin, all on GPU. Distinguishes kernel bugs
from inherent bf16 error amplification."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import torch.nn.functional as F
from pytorch_distributed_train_amd.models import build_model
from pytorch_distributed_train_amd.ops import functional as OF
from tests.torch_twin import twin_resnet18

DEV = "cuda:0"
torch.manual_seed(2)
ours = build_model("resnet18", 10, torch.bfloat16).to(DEV)
sd = {k: v.clone() for k, v in ours.state_dict().items()}

twin32 = twin_resnet18(10).to(DEV)
twin32.load_state_dict(sd)
twin_ac = twin_resnet18(10).to(DEV)
twin_ac.load_state_dict(sd)

x = torch.randn(8, 3, 64, 64, device=DEV)
labels = torch.randint(0, 10, (8,), device=DEV)

ours.train(); twin32.train(); twin_ac.train()
OF.softmax_cross_entropy(ours(x), labels).backward()
F.cross_entropy(twin32(x), labels).backward()
with torch.autocast("cuda", torch.bfloat16):
    loss = F.cross_entropy(twin_ac(x), labels)
loss.backward()

g_o = {n: p.grad.float() for n, p in ours.named_parameters()}
g_32 = {n: p.grad.float() for n, p in twin32.named_parameters()}
g_ac = {n: p.grad.float() for n, p in twin_ac.named_parameters()}

print(f"{'param':35s} {'|ref|max':>9s} {'ours_err':>9s} {'ac_err':>9s}")
for n in g_32:
    r = g_32[n].abs().max().item()
    eo = (g_o[n] - g_32[n]).abs().max().item()
    ea = (g_ac[n] - g_32[n]).abs().max().item()
    flag = " <<<" if eo > 3 * max(ea, 1e-4) and eo > 0.05 * max(r, 0.1) else ""
    print(f"{n:35s} {r:9.4f} {eo:9.4f} {ea:9.4f}{flag}")
