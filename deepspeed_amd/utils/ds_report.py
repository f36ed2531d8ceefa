"""Environment report CLI (reference: bin/ds_report ->
deepspeed/env_report.py): shows torch/ROCm versions, op availability,
and GPU topology. `python -m deepspeed_amd.utils.ds_report`."""

import shutil
import subprocess
import sys


def _row(k, v):
    print(f"{k:.<40s} {v}")


def main():
    import torch
    import deepspeed_amd
    print("-" * 60)
    print("deepspeed_amd environment report")
    print("-" * 60)
    _row("deepspeed_amd version", deepspeed_amd.__version__)
    _row("torch version", torch.__version__)
    _row("torch hip version", getattr(torch.version, "hip", None) or "n/a")
    _row("GPU available", str(torch.cuda.is_available()))
    if torch.cuda.is_available():
        _row("device count", str(torch.cuda.device_count()))
        _row("device name", torch.cuda.get_device_name(0))
        _row("gcn arch", torch.cuda.get_device_properties(0).gcnArchName)
    _row("hipcc", shutil.which("hipcc") or "NOT FOUND")

    print("-" * 60)
    print("op availability")
    print("-" * 60)
    from deepspeed_amd.ops._loader import get_ext
    ext = get_ext()
    if ext is None:
        _row("native extension", "NOT BUILT (python setup.py build_ext "
             "--inplace)")
    else:
        for op in ("fused_adam_flat", "cpu_adam_flat", "norm_fwd", "norm_bwd",
                   "rope", "gated_act_fwd", "gated_act_bwd", "groupwise_quant",
                   "groupwise_dequant", "AioHandle"):
            _row(op, "OK" if hasattr(ext, op) else "MISSING")
    if torch.cuda.is_available() and shutil.which("rocm-smi"):
        print("-" * 60)
        subprocess.run(["rocm-smi", "--showtopo"], check=False)
    return 0


if __name__ == "__main__":
    sys.exit(main())
