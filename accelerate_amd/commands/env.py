"""`accelerate-amd env` — platform report (reference: commands/env.py)."""

import platform
import subprocess

import torch

import accelerate_amd


def env_command(args):
    info = {
        "accelerate_amd version": accelerate_amd.__version__,
        "Platform": platform.platform(),
        "Python version": platform.python_version(),
        "PyTorch version": torch.__version__,
        "ROCm (HIP) version": getattr(torch.version, "hip", None),
        "GPU available": torch.cuda.is_available(),
        "GPU count": torch.cuda.device_count() if torch.cuda.is_available() else 0,
    }
    if torch.cuda.is_available():
        props = torch.cuda.get_device_properties(0)
        info["GPU 0"] = f"{props.name} ({props.gcnArchName}), {props.total_memory / 2**30:.0f} GiB HBM"
    try:
        from accelerate_amd.ops import has_extension

        info["HIP kernel pack (accelerate_amd._C)"] = "built" if has_extension() else "NOT built"
    except Exception:
        info["HIP kernel pack (accelerate_amd._C)"] = "NOT built"
    try:
        smi = subprocess.run(["rocm-smi", "--showproductname"], capture_output=True, text=True, timeout=10)
        if smi.returncode == 0:
            info["rocm-smi"] = "available"
    except Exception:
        pass
    print("\n- " + "\n- ".join(f"{k}: {v}" for k, v in info.items()) + "\n")


def add_parser(subparsers):
    parser = subparsers.add_parser("env", help="Print environment information")
    parser.set_defaults(func=env_command)
    return parser
