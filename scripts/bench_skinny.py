import sys, os, time, torch
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
from tensorlink_amd import ops

def timeit(fn, iters=30):
    for _ in range(5): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6  # us

shapes = [(256,4608,3584,"qkv"),(256,3584,3584,"o"),(256,18944,3584,"gateup"),(256,3584,18944,"down"),(16,3584,3584,"o_m16")]
for M,N,K,name in shapes:
    x = torch.randn(M,K,device="cuda",dtype=torch.bfloat16)
    w = torch.randn(N,K,device="cuda",dtype=torch.bfloat16)/K**0.5
    t_lib = timeit(lambda: torch.nn.functional.linear(x,w))
    t_sk  = timeit(lambda: ops._require_ext().skinny_gemm(x,w,None))
    wbytes = N*K*2/1e9
    print(f"{name:8s} M{M} N{N} K{K}: lib {t_lib:7.1f}us ({wbytes/t_lib*1e3:6.2f} TB/s)  skinny {t_sk:7.1f}us ({wbytes/t_sk*1e3:6.2f} TB/s)")
