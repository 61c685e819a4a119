"""A/B the StaggerU K walk (TD_GEMM_STAGGER) on the ring + split-K
tiers: numerics vs fp32 ref, then TF at the headline GEMM shapes."""
import os
import subprocess
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def body():
    import torch

    from triton_dist_amd import _C

    s = torch.cuda.current_stream().cuda_stream

    def t(fn, n=30):
        for _ in range(5):
            fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(n):
            fn()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / n

    for m, n, k in [(4096, 4096, 4096), (8192, 8192, 8192),
                    (512, 5120, 25600), (512, 5120, 27648)]:
        a = (torch.randn(m, k, device="cuda") / 8).to(torch.bfloat16)
        w = (torch.randn(n, k, device="cuda") / 8).to(torch.bfloat16)
        c = torch.empty(m, n, device="cuda", dtype=torch.bfloat16)
        if m >= 4096:
            _C.gemm_bf16(a.data_ptr(), w.data_ptr(), c.data_ptr(), 0,
                         m, n, k, s)
            torch.cuda.synchronize()
            ref = a.float() @ w.float().t()
            rel = ((c.float() - ref).abs().max() / ref.abs().max()).item()
            sec = t(lambda: _C.gemm_bf16(a.data_ptr(), w.data_ptr(),
                                         c.data_ptr(), 0, m, n, k, s))
            tf = 2 * m * n * k / sec / 1e12
            print(f"ring {m}x{n}x{k}: {tf:7.1f} TF rel {rel:.1e}",
                  flush=True)
        else:
            from triton_dist_amd.ops.gemm import sk256_pick
            sk = sk256_pick(m, n, k)
            ws = torch.empty(sk, m, n, dtype=torch.float32, device="cuda")
            _C.gemm256_sk2_bf16(a.data_ptr(), w.data_ptr(), c.data_ptr(),
                                0, ws.data_ptr(), m, n, k, sk, s)
            torch.cuda.synchronize()
            ref = a.float() @ w.float().t()
            rel = ((c.float() - ref).abs().max() / ref.abs().max()).item()
            sec = t(lambda: _C.gemm256_sk2_bf16(
                a.data_ptr(), w.data_ptr(), c.data_ptr(), 0,
                ws.data_ptr(), m, n, k, sk, s), n=50)
            print(f"sk2  {m}x{n}x{k} (sk{sk}): {sec*1e6:6.1f} us "
                  f"rel {rel:.1e}", flush=True)


if __name__ == "__main__":
    if len(sys.argv) > 1 and sys.argv[1] == "body":
        body()
    else:
        for v in ("1", "0"):
            print(f"== TD_GEMM_STAGGER={v}")
            subprocess.run([sys.executable, __file__, "body"],
                           env={**os.environ, "TD_GEMM_STAGGER": v},
                           timeout=240)
