import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

import modal_examples_amd.ops.functional as F

q = torch.randn(4, 10, 4096, 64, device="cuda", dtype=torch.bfloat16)
k = torch.randn_like(q)
v = torch.randn_like(q)
for _ in range(5):
    F.attention(q, k, v)
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(30):
    F.attention(q, k, v)
torch.cuda.synchronize()
dt = (time.perf_counter() - t0) / 30
print(f"ABL={os.environ.get('MODAL_AMD_FA_ABLATE', '0')}: {dt*1e3:.3f} ms")
