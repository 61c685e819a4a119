"""Environment diagnostics: `python -m triton_dist_amd.doctor`.

One screen that answers "is this box ready for the framework?" —
versions, devices, xGMI mesh, P2P attributes, extension build state,
and a heap + kernel smoke when a GPU is present. The counterpart of the
reference's platform-detect utilities (utils.py:51-114, amd_utils.py),
packaged as a command.
"""
from __future__ import annotations

import sys


def main() -> int:
    import torch

    print("== triton_dist_amd doctor ==")
    print(f"python {sys.version.split()[0]}  torch {torch.__version__}")
    hip = getattr(torch.version, "hip", None)
    print(f"hip {hip}  cuda.is_available {torch.cuda.is_available()}")

    from . import _C

    print(f"extension: {'LOADED' if _C is not None else 'NOT BUILT'}"
          f"{'' if _C is None else ' (' + _C.__file__ + ')'}")

    if not torch.cuda.is_available():
        print("no GPU: CPU/gloo mock-heap mode only")
        return 0

    n = torch.cuda.device_count()
    for d in range(n):
        p = torch.cuda.get_device_properties(d)
        print(f"gpu{d}: {p.name}  {p.total_memory / 1e9:.0f} GB  "
              f"{p.multi_processor_count} CUs  arch {p.gcnArchName}")

    from .utils.device_info import is_full_mesh_xgmi, p2p_attributes

    if n > 1:
        try:
            print(f"full-mesh xGMI: {is_full_mesh_xgmi()}")
        except Exception as e:  # rocm-smi may be absent
            print(f"xGMI topology query failed: {e}")
        for d in range(min(n, 2)):
            for peer in range(n):
                if peer != d:
                    print(f"p2p {d}->{peer}: {p2p_attributes(d, peer)}")

    # heap + kernel smoke (single process)
    try:
        from .runtime.symm_mem import SymmHeap

        heap = SymmHeap(size_mb=64)
        buf = heap.alloc_buffer((128,), torch.int32)
        buf.local().fill_(7)
        torch.cuda.synchronize()
        assert int(buf.local().sum()) == 7 * 128
        heap.close()
        print("symm heap: OK (alloc + fill + read back)")
    except Exception as e:
        print(f"symm heap: FAILED — {e}")
        return 1

    try:
        from .ops.fused import swiglu_op

        h = torch.randn(64, 256, device="cuda").to(torch.bfloat16)
        swiglu_op(h, 128)
        torch.cuda.synchronize()
        print("HIP kernel smoke (swiglu): OK")
    except Exception as e:
        print(f"HIP kernel smoke: FAILED — {e}")
        return 1
    print("all checks passed")
    return 0


if __name__ == "__main__":
    sys.exit(main())
