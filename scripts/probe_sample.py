import sys, torch
sys.path.insert(0, "/root/repo")
from tensorlink_amd import ops
C = ops._require_ext()
dev = "cuda:0"
torch.manual_seed(1)
B, V = 2, 1024
logits = torch.randn(B, V, device=dev, dtype=torch.bfloat16)
p = dict(temps=torch.full((B,), 0.8, device=dev),
         top_ps=torch.full((B,), 0.9, device=dev),
         top_ks=torch.zeros(B, device=dev, dtype=torch.int32),
         pres=torch.zeros(B, device=dev), freqs=torch.zeros(B, device=dev))
ctr = torch.zeros(1, device=dev, dtype=torch.int64)
# eager determinism with counter
seq1 = []
ctr.zero_()
for _ in range(6):
    C.bump_sample_counter(ctr)
    seq1.append(ops.sample_tokens(logits, counter=ctr, seed_base=5, **p).tolist())
seq2 = []
ctr.zero_()
for _ in range(6):
    C.bump_sample_counter(ctr)
    seq2.append(ops.sample_tokens(logits, counter=ctr, seed_base=5, **p).tolist())
print("eager deterministic:", seq1 == seq2, seq1[:3], seq2[:3])

# graph determinism
out = torch.zeros(B, dtype=torch.int64, device=dev)
s = torch.cuda.Stream(); s.wait_stream(torch.cuda.current_stream())
def step():
    C.bump_sample_counter(ctr)
    out.copy_(ops.sample_tokens(logits, counter=ctr, seed_base=5, **p))
with torch.cuda.stream(s):
    step()
torch.cuda.current_stream().wait_stream(s)
g = torch.cuda.CUDAGraph()
with torch.cuda.graph(g):
    step()
r1 = []
ctr.zero_()
for _ in range(6):
    g.replay(); r1.append(out.tolist())
print("ctr after r1:", int(ctr[0]))
r2 = []
ctr.zero_()
for _ in range(6):
    g.replay(); r2.append(out.tolist())
print("graph deterministic:", r1 == r2, r1[:3], r2[:3])
print("graph==eager:", r1 == seq1)
