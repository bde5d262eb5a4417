import sys, os, time, torch
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
from tensorlink_amd import ops

def timeit(fn, iters=20):
    for _ in range(3): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3  # ms

for B,S,Hq,Hkv,D,name in [(256,512,28,4,128,"qwen7b-prefill"),(8,4096,32,8,128,"long4k"),(1,8192,32,8,128,"long8k")]:
    q = torch.randn(B,S,Hq,D,device="cuda",dtype=torch.bfloat16)
    k = torch.randn(B,S,Hkv,D,device="cuda",dtype=torch.bfloat16)
    v = torch.randn(B,S,Hkv,D,device="cuda",dtype=torch.bfloat16)
    t_mine = timeit(lambda: ops.attention_prefill(q,k,v))
    rep = Hq//Hkv
    qt,kt,vt = q.transpose(1,2), k.transpose(1,2).repeat_interleave(rep,1), v.transpose(1,2).repeat_interleave(rep,1)
    t_sdpa = timeit(lambda: torch.nn.functional.scaled_dot_product_attention(qt,kt,vt,is_causal=True))
    flops = 2*2*B*Hq*S*S*D/2
    print(f"{name}: mine {t_mine:.2f}ms ({flops/t_mine/1e9:.0f} TF)  sdpa {t_sdpa:.2f}ms ({flops/t_sdpa/1e9:.0f} TF)")
