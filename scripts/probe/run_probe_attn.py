import ctypes, time, torch
lib = ctypes.CDLL("scripts/probe/_probe_attn.so")
torch.manual_seed(0)
B, qh, kvh, D, maxlen = 512, 64, 8, 128, 1024
q = torch.randn(B, qh, D, device="cuda").to(torch.bfloat16)
k = torch.randn(B, maxlen, kvh, D, device="cuda").to(torch.bfloat16)
v = torch.randn(B, maxlen, kvh, D, device="cuda").to(torch.bfloat16)
out = torch.empty_like(q)
stream = ctypes.c_void_p(torch.cuda.current_stream().cuda_stream)
def run(var, off_t):
    lib.run_fd(var, ctypes.c_void_p(q.data_ptr()), ctypes.c_void_p(k.data_ptr()),
               ctypes.c_void_p(v.data_ptr()), ctypes.c_void_p(out.data_ptr()),
               ctypes.c_void_p(off_t.data_ptr()), B, qh, kvh, maxlen, stream)
tro = torch.zeros(4*64*4, device="cuda", dtype=torch.float32)
lib.run_trprobe(ctypes.c_void_p(tro.data_ptr()), stream)
torch.cuda.synchronize()
got = tro.view(4, 64, 4).cpu().int()
names = ["all-0", "(l&3)*8B", "(l>>4)*128B", "l*8B"]
for p in range(4):
    print(f"pattern {names[p]}:")
    for l in range(0, 64, 1):
        if l in (0,1,2,3,4,5,16,17,20,32,48,60,63):
            print(f"  lane{l:2d}: {got[p,l].tolist()}")

for L in (131, 512, 1024):
    off = torch.tensor([L - 1], dtype=torch.int64, device="cuda")
    print(f"-- L={L} (KV {B*L*kvh*D*2*2/1e6:.0f} MB)")
    run(0, off); torch.cuda.synchronize(); ref = out.clone()
    run(4, off); torch.cuda.synchronize()
    rel = (out.float()-ref.float()).abs().max().item() / ref.float().abs().max().item()
    print(f"  VAR4 vs VAR0 max rel err: {rel:.2e}")
    for var, name in [(0,"full"),(4,"PV-MFMA"),(1,"no-PV")]:
        for _ in range(5): run(var, off)
        torch.cuda.synchronize(); t0=time.perf_counter()
        for _ in range(50): run(var, off)
        torch.cuda.synchronize()
        us=(time.perf_counter()-t0)/50*1e6
        print(f"  {name:14s} {us:7.1f} us")
