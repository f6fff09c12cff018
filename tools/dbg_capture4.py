import os, sys
sys.path.insert(0, os.environ.get("GRAFT_REPO_ROOT", "/root/repo"))
import numpy as np
import torch
import flox_amd

g = torch.Generator(device="cuda").manual_seed(1)
arr = torch.rand((1000, 720, 1440), generator=g, dtype=torch.float32, device="cuda")
hours = (torch.arange(1000, device="cuda") % 24).to(torch.int64)
view = arr.permute(1, 2, 0)
eg = np.arange(24)

def step():
    res, _ = flox_amd.groupby_reduce(view, hours, func="mean", expected_groups=eg)
    return res

step(); torch.cuda.synchronize()
# probe 1: torch.sort under capture
try:
    gr = torch.cuda.CUDAGraph()
    with torch.cuda.graph(gr):
        s, p = torch.sort(hours, stable=True)
    gr.replay(); torch.cuda.synchronize()
    print("torch.sort captures OK")
except Exception as e:
    print("torch.sort capture FAIL:", type(e).__name__, str(e)[:200])
# probe 2: full step
try:
    gr2 = torch.cuda.CUDAGraph()
    with torch.cuda.graph(gr2):
        r = step()
    gr2.replay(); torch.cuda.synchronize()
    print("full cols step captures OK")
except Exception as e:
    import traceback
    print("full step capture FAIL:", type(e).__name__, str(e)[:300])
    tb = traceback.format_exc().splitlines()
    print("\n".join(tb[-12:]))
