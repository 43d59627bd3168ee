"""`python -m lzy_amd` — environment doctor.

Prints what a support thread would ask for: versions, native component
status, GPU visibility, resolved config.
"""
from __future__ import annotations

import json
import sys


def main() -> int:
    import torch

    import lzy_amd
    from lzy_amd import ops
    from lzy_amd.config import get_config
    from lzy_amd.sched import NATIVE as SCHED_NATIVE

    cfg = get_config()
    report = {
        "lzy_amd": lzy_amd.__version__,
        "python": sys.version.split()[0],
        "torch": torch.__version__,
        "cuda_available": torch.cuda.is_available(),
        "gpus": torch.cuda.device_count() if torch.cuda.is_available() else 0,
        "gpu_name": (
            torch.cuda.get_device_name(0) if torch.cuda.is_available() else None
        ),
        "native": {
            "sched_core_cpp": SCHED_NATIVE,
            "hip_ops": ops.NATIVE,
        },
        "config": {
            k: getattr(cfg, k)
            for k in (
                "storage", "channel_transport", "channel_chunk_mb",
                "channel_wire_cast", "exec_threads", "chain_dispatch",
                "op_streams", "cache_enabled", "spill_enabled",
            )
        },
    }
    print(json.dumps(report, indent=2))
    ok = SCHED_NATIVE and (ops.NATIVE or not torch.cuda.is_available())
    if not ok:
        print("WARNING: native components missing — run "
              "`python setup.py build_ext --inplace`", file=sys.stderr)
    return 0 if ok else 1


if __name__ == "__main__":
    sys.exit(main())
