"""rocprofv3 probe: tuned bf16 median kernels vs generic rsel engine at
n=256, d=8M (same data)."""
import sys, os
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch
from byzpy_amd.hip import dispatch as D

def main():
    X = torch.empty(256, 8_000_000, dtype=torch.bfloat16, device="cuda").normal_()
    Xf = X.float()
    for _ in range(3):
        D.median(X)           # generic engine (d%4==0 routes here)
        D.trimmed_mean(X, 32) # generic: 2 levels + sum
        D.median(Xf)          # generic f32: 4 levels + finalize
    torch.cuda.synchronize()

if __name__ == "__main__":
    main()
