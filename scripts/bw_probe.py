import torch, time
def t(fn, n=20):
    for _ in range(3): fn()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/n
a = torch.randn(1_250_000, 192, device="cuda")            # 960 MB f32
b = torch.empty_like(a)
c = torch.empty(128, 1_250_000, device="cuda")            # 640 MB f32
s = t(lambda: torch.sum(a))                               # pure read
cp = t(lambda: b.copy_(a))                                # read+write
f = t(lambda: c.fill_(1.0))                               # pure write
GB = 1e9
print(f"read 960MB: {s*1e6:.0f} us = {a.numel()*4/s/1e12:.2f} TB/s")
print(f"copy 960MB: {cp*1e6:.0f} us = {2*a.numel()*4/cp/1e12:.2f} TB/s (rd+wr)")
print(f"write 640MB: {f*1e6:.0f} us = {c.numel()*4/f/1e12:.2f} TB/s")
