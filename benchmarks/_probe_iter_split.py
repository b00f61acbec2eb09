"""Profile probe: per-kernel split of the fused Weiszfeld / CC iterations
(dist pass vs update pass) at the headline shape. Run under rocprofv3."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

from byzpy_amd.hip import require

ext = require()
n, d = 64, 125_000_000
X = torch.empty(n, d, dtype=torch.bfloat16, device="cuda").normal_()
z = torch.zeros(d, dtype=torch.float32, device="cuda")
sh = torch.zeros((), dtype=torch.float32, device="cuda")
for _ in range(10):
    ext.weiszfeld_iter(X, z, 1e-12, sh)
for _ in range(10):
    ext.cc_iter(X, z, 0.5, 1e-12)
torch.cuda.synchronize()

# CAF at suite shape: 10.3 ms for 64x65536 is launch/sync bound -- trace it
from byzpy_amd.hip import dispatch as D

Xs = torch.empty(64, 65536, dtype=torch.bfloat16, device="cuda").normal_()
for _ in range(5):
    D.caf(Xs, 8)
torch.cuda.synchronize()
print("done")
