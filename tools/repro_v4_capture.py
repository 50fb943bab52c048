import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import deepfake_detection_amd as dfd
from deepfake_detection_amd.optim import RMSpropTF

which = sys.argv[1] if len(sys.argv) > 1 else "full"
model_name = sys.argv[2] if len(sys.argv) > 2 else "efficientnet_deepfake_v4"
opt_name = sys.argv[3] if len(sys.argv) > 3 else "rmsproptf"
in_ch = 12 if "v4" in model_name else 3
m = dfd.create_model(model_name, num_classes=2, in_chans=in_ch).cuda().to(memory_format=torch.channels_last)
if opt_name == "sgd":
    opt = torch.optim.SGD(m.parameters(), lr=1e-5)
else:
    opt = RMSpropTF(m.parameters(), lr=1e-5, alpha=0.9, eps=1e-3, momentum=0.9)
x = torch.randn(4, in_ch, 128, 128, device="cuda").contiguous(memory_format=torch.channels_last)
t = torch.randint(0, 2, (4,), device="cuda")

def fwd():
    with torch.autocast("cuda", torch.bfloat16):
        return torch.nn.functional.cross_entropy(m(x), t)

def step(with_bwd=True, with_opt=True):
    loss = fwd()
    if with_bwd:
        opt.zero_grad(set_to_none=True)
        loss.backward()
    if with_opt:
        opt.step()

for _ in range(3):
    step()
torch.cuda.synchronize()
g = torch.cuda.CUDAGraph()
try:
    with torch.cuda.graph(g):
        if which == "fwd":
            fwd()
        elif which == "bwd":
            step(with_opt=False)
        else:
            step()
    torch.cuda.synchronize()
    print("CAPTURE OK:", which, model_name, opt_name)
except Exception as e:
    print("CAPTURE FAIL:", which, model_name, opt_name, "->", str(e)[:160])
