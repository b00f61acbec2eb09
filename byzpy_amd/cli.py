"""byzpy-amd CLI: version / doctor / list.

Reference parity: byzpy/cli.py:122-158 (doctor probes the runtime;
list walks operator subclasses). The doctor here probes the ROCm stack:
torch-ROCm, GPU visibility, RCCL backend, and the in-tree HIP extension.
"""
from __future__ import annotations

import argparse
import json
import sys


def _doctor() -> dict:
    info: dict = {}
    try:
        import torch

        info["torch"] = torch.__version__
        info["hip"] = getattr(torch.version, "hip", None)
        info["cuda_available"] = torch.cuda.is_available()
        if torch.cuda.is_available():
            info["device_count"] = torch.cuda.device_count()
            info["device_name"] = torch.cuda.get_device_name(0)
        info["nccl_rccl_backend"] = torch.distributed.is_nccl_available()
        info["gloo_backend"] = torch.distributed.is_gloo_available()
    except Exception as e:  # noqa: BLE001
        info["torch_error"] = repr(e)
    from byzpy_amd import hip

    info["hip_extension"] = hip.available()
    from byzpy_amd._version import __version__

    info["byzpy_amd"] = __version__
    return info


def _walk_subclasses(base) -> list:
    out = []
    stack = [base]
    while stack:
        cls = stack.pop()
        for sub in cls.__subclasses__():
            stack.append(sub)
            if not sub.__name__.startswith("_"):
                out.append(sub)
    return sorted({c.__name__ for c in out})


def main(argv=None) -> int:
    parser = argparse.ArgumentParser(prog="byzpy-amd")
    sub = parser.add_subparsers(dest="cmd")
    sub.add_parser("version")
    doc = sub.add_parser("doctor")
    doc.add_argument("--format", choices=["text", "json"], default="text")
    lst = sub.add_parser("list")
    lst.add_argument(
        "kind", choices=["aggregators", "attacks", "pre-aggregators"]
    )
    args = parser.parse_args(argv)

    if args.cmd == "version":
        from byzpy_amd._version import __version__

        print(__version__)
        return 0
    if args.cmd == "doctor":
        info = _doctor()
        if args.format == "json":
            print(json.dumps(info, indent=2, default=str))
        else:
            for k, v in info.items():
                print(f"{k}: {v}")
        return 0
    if args.cmd == "list":
        import byzpy_amd.aggregators  # noqa: F401
        import byzpy_amd.attacks  # noqa: F401
        import byzpy_amd.pre_aggregators  # noqa: F401
        from byzpy_amd.aggregators.base import Aggregator
        from byzpy_amd.attacks.base import Attack
        from byzpy_amd.pre_aggregators.base import PreAggregator

        base = {
            "aggregators": Aggregator,
            "attacks": Attack,
            "pre-aggregators": PreAggregator,
        }[args.kind]
        for name in _walk_subclasses(base):
            print(name)
        return 0
    parser.print_help()
    return 1


if __name__ == "__main__":
    sys.exit(main())
