import os, sys, torch
sys.path.insert(0, "/root/repo")
from distributedmnist_amd import _C
ext = _C.ext()
B = 8192
y1 = torch.randn(B,14,14,32, device="cuda").bfloat16().contiguous()
dact2 = torch.randn(B,14,14,64, device="cuda").bfloat16().contiguous()
dw = torch.zeros(5,5,32,64, device="cuda")
def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup): fn()
    torch.cuda.synchronize()
    e0, e1 = torch.cuda.Event(True), torch.cuda.Event(True)
    ts=[]
    for _ in range(iters):
        e0.record(); fn(); e1.record(); e1.synchronize()
        ts.append(e0.elapsed_time(e1)*1000)
    ts.sort(); return ts[len(ts)//2]
print("ABL", os.environ.get("DMNIST_DW_ABL","0"), "->",
      f"{timeit(lambda: ext.conv_dw_into(y1, dact2, dw)):.1f} us")
