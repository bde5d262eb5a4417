import sys, torch
sys.path.insert(0, "/root/repo")
from tensorlink_amd import ops
from tensorlink_amd.parallel.pipeline import PipelineRunner, SamplingParams
from tensorlink_amd.parallel.planner import plan_for_world
C = ops._require_ext()
dev = torch.device("cuda", 0)
r = PipelineRunner(plan_for_world("tiny", 1), 0, 1, device=dev,
                   dtype=torch.bfloat16, seed=11)
torch.manual_seed(45)
ids = torch.randint(0, 1024, (2, 16)).to(dev)
# prefill once to set up a cache
cache = r.alloc_cache(2, 40)
pos = torch.arange(16, device=dev, dtype=torch.int32).unsqueeze(0).expand(2, -1).contiguous()
with torch.no_grad():
    logits = r.stage(ids, pos, kv_cache=cache)
t0 = logits[:, -1].argmax(-1)
lens0 = cache.seq_lens.clone()
B = 2
params = dict(temps=torch.full((B,), 0.8, device=dev),
              top_ps=torch.full((B,), 0.9, device=dev),
              top_ks=torch.zeros(B, device=dev, dtype=torch.int32),
              pres=torch.zeros(B, device=dev), freqs=torch.zeros(B, device=dev))
ctr = torch.zeros(1, device=dev, dtype=torch.int64)
tok_buf = t0.clone()
pos_buf = torch.full((B,), 16, device=dev, dtype=torch.int32)
dbg = torch.zeros(B, device=dev)

@torch.no_grad()
def step():
    C.bump_sample_counter(ctr)
    lg = r.stage(tok_buf.unsqueeze(1), pos_buf.unsqueeze(1), kv_cache=cache).squeeze(1)
    dbg.copy_(lg.float().abs().sum(-1))
    tok_buf.copy_(ops.sample_tokens(lg, counter=ctr, seed_base=5, **params))
    pos_buf.add_(1)

s = torch.cuda.Stream(); s.wait_stream(torch.cuda.current_stream())
with torch.cuda.stream(s):
    step()
torch.cuda.current_stream().wait_stream(s)
cache.seq_lens.copy_(lens0)
g = torch.cuda.CUDAGraph()
with torch.cuda.graph(g):
    step()
cache.seq_lens.copy_(lens0)
for call in range(3):
    cache.seq_lens.copy_(lens0)
    tok_buf.copy_(t0)
    pos_buf.fill_(16)
    ctr.zero_()
    rec = []
    for t in range(3):
        g.replay()
        rec.append((round(float(dbg[0]), 2), int(tok_buf[0])))
    print("call", call, rec)
