"""`python -m dcr_amd` — build/environment/kernel status report."""
import hashlib
import platform

import torch

import dcr_amd
from dcr_amd import ops


def main():
    print(f"dcr_amd {dcr_amd.__version__}")
    print(f"torch {torch.__version__} (hip {torch.version.hip})")
    print(f"python {platform.python_version()} on {platform.machine()}")
    print(f"cuda/hip device available: {torch.cuda.is_available()}")
    m = ops.ext()
    if m is None:
        print("HIP extension: NOT BUILT (run __graft_entry__.build())")
    else:
        kernels = sorted(n for n in dir(m) if not n.startswith("_"))
        from pathlib import Path
        so = Path(ops.__file__).parent / "_dcr_hip.so"
        digest = hashlib.sha256(so.read_bytes()).hexdigest()[:16]
        print(f"HIP extension: {so} (sha256 {digest})")
        print(f"  exported ops ({len(kernels)}): {', '.join(kernels)}")
    if ops.dispatch_counts:
        print(f"dispatch counts: {dict(ops.dispatch_counts)}")
    import os
    gates = ["DCR_NATIVE_CONV", "DCR_NATIVE_CONV_BWD", "DCR_NATIVE_GEMM",
             "DCR_ATTN_V4", "DCR_ATTN_BWD_V4", "DCR_HIPGRAPH",
             "DCR_ATTN_V2", "DCR_DEV_ADAMW", "DCR_PROFILE",
             "DCR_AMD_ALLOW_FALLBACK"]
    active = {g: os.environ[g] for g in gates if g in os.environ}
    desc = active if active else "defaults (native conv on; round-2 drafts off)"
    print(f"env gates: {desc}")


if __name__ == "__main__":
    main()
