"""PMC probe for round-2 kernels: small-n Gram (MFMA evidence) and the
radix level passes (occupancy/wave evidence). Times under PMC are NOT
wall-clock."""
import sys, os
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch
from byzpy_amd.hip import dispatch as D

def main():
    g = torch.Generator().manual_seed(0)
    X8 = torch.randn(8, 50_000_000, generator=g).to("cuda", torch.bfloat16)
    for _ in range(3):
        D.gram(X8)  # gram_bf16_small16_kernel (P=2 packing)
    X512 = torch.empty(512, 4_000_000, dtype=torch.bfloat16, device="cuda").normal_()
    for _ in range(2):
        D.median(X512)        # rsel 3 levels + out
        D.trimmed_mean(X512, 64)
    torch.cuda.synchronize()

if __name__ == "__main__":
    main()
