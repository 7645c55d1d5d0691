import torch, time, numpy as np
from concurrent.futures import ThreadPoolExecutor
ts = [torch.randn(3,224,224).to(torch.bfloat16) for _ in range(64)]
pin = torch.empty(64,3,224,224, dtype=torch.bfloat16, pin_memory=True)
def t(fn, n=30):
    for _ in range(5): fn()
    t0=time.perf_counter()
    for _ in range(n): fn()
    return (time.perf_counter()-t0)/n*1000
print("stack         : %.2f ms" % t(lambda: torch.stack(ts, dim=0, out=pin)))
def copyloop():
    for i, x in enumerate(ts): pin[i].copy_(x)
print("copy_ loop    : %.2f ms" % t(copyloop))
pv = pin.view(torch.int16)
tv = [x.view(torch.int16) for x in ts]
def copyloop16():
    for i, x in enumerate(tv): pv[i].copy_(x)
print("int16 view    : %.2f ms" % t(copyloop16))
pnp = pin.view(torch.int16).numpy()
tnp = [x.view(torch.int16).numpy() for x in ts]
def nploop():
    for i, x in enumerate(tnp): pnp[i] = x
print("numpy assign  : %.2f ms" % t(nploop))
ex = ThreadPoolExecutor(8)
def npar(nthreads=8):
    chunk = 64 // nthreads
    def cp(s):
        for i in range(s, s+chunk): pnp[i] = tnp[i]
    futs = [ex.submit(cp, i*chunk) for i in range(nthreads)]
    for f in futs: f.result()
print("numpy par8    : %.2f ms" % t(npar))
flat = torch.cat([x.reshape(1,-1) for x in ts], dim=0)
print("cat reshape   : %.2f ms" % t(lambda: torch.cat([x.reshape(1,-1) for x in ts], dim=0)))
import ctypes
dst = pnp.ctypes.data
n_bytes = tnp[0].nbytes
def memmove_loop():
    for i, x in enumerate(tnp):
        ctypes.memmove(dst + i*n_bytes, x.ctypes.data, n_bytes)
print("ctypes memmove: %.2f ms" % t(memmove_loop))
