"""Decode-shape GEMM tier comparison: hipBLASLt vs dispatch vs gemm256_sk."""
import torch, time
from triton_dist_amd.ops.gemm import gemm, choose_splits
from triton_dist_amd import _C

def t(fn, n=50):
    for _ in range(5): fn()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize()
    return (time.perf_counter()-t0)/n*1e6

stream = torch.cuda.current_stream().cuda_stream
shapes = [("qkv",512,10240,5120,(2,4,5,8,10)),("o",512,5120,8192,(2,4,8,16)),
          ("gate_up",512,51200,5120,(2,4)),("down",512,5120,25600,(4,5,8,10,25))]
for name,m,n,k,sks in shapes:
    a = torch.randn(m,k,device="cuda").to(torch.bfloat16)
    w = torch.randn(n,k,device="cuda").to(torch.bfloat16)
    c = torch.empty(m,n,device="cuda",dtype=torch.bfloat16)
    ws = torch.empty(m,n,device="cuda",dtype=torch.float32)
    ref = torch.matmul(a, w.t())
    gf = 2*m*n*k/1e9
    us_blt = t(lambda: torch.matmul(a, w.t()))
    line = f"{name:8s}: blt {us_blt:6.1f}us ({gf/us_blt*1e3:5.0f} TF)"
    best = (us_blt, "blt")
    for sk in sks:
        if k % (128*sk): continue
        _C.gemm256_sk_bf16(a.data_ptr(), w.data_ptr(), c.data_ptr(), 0,
                           ws.data_ptr(), m, n, k, sk, stream)
        torch.cuda.synchronize()
        err = (c.float()-ref.float()).abs().max().item()
        rel = err / ref.float().abs().max().item()
        us = t(lambda: _C.gemm256_sk_bf16(a.data_ptr(), w.data_ptr(),
               c.data_ptr(), 0, ws.data_ptr(), m, n, k, sk, stream))
        line += f" | sk{sk} {us:6.1f} ({gf/us*1e3:5.0f} TF, rel {rel:.1e})"
        if us < best[0]: best = (us, f"sk{sk}")
    print(line)
    print(f"  -> best: {best[1]} {best[0]:.1f}us")
