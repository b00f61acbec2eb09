"""roctx marker evidence probe: a PS round with trace_range markers
around the aggregate and the kernel calls (rocprofv3 --marker-trace)."""
import sys, os
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch
from byzpy_amd.hip import dispatch as D
from byzpy_amd.utils.tracing import enabled, trace_range

def main():
    print("roctx enabled:", enabled())
    X = torch.randn(32, 1_000_000).to("cuda", torch.bfloat16)
    for r in range(3):
        with trace_range(f"ps_round_{r}"):
            with trace_range("median"):
                D.median(X)
            with trace_range("multi_krum"):
                D.multi_krum(X, 8, 6)
            with trace_range("geomed_fixed"):
                D.geometric_median(X, fixed_iters=8)
    torch.cuda.synchronize()

if __name__ == "__main__":
    main()
