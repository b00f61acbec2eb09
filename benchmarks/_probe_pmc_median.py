"""PMC probe for the selection-network median (round-2 late): VALU
instruction count + wave count at the bench shape. Times under PMC are
NOT wall-clock."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch

from byzpy_amd.hip import dispatch as D


def main():
    X = torch.empty(64, 125_000_000, dtype=torch.bfloat16, device="cuda")
    X.normal_()
    for _ in range(2):
        D.median(X)
    torch.cuda.synchronize()


if __name__ == "__main__":
    main()
