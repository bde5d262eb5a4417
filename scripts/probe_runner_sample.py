import sys, torch
sys.path.insert(0, "/root/repo")
from tensorlink_amd.parallel.pipeline import PipelineRunner, SamplingParams
from tensorlink_amd.parallel.planner import plan_for_world
dev = torch.device("cuda", 0)
r = PipelineRunner(plan_for_world("tiny", 1), 0, 1, device=dev,
                   dtype=torch.bfloat16, seed=11)
orig = PipelineRunner._graph_decode
def patched(self, cur, positions, out_tokens, T, sp):
    c = self.kv_cache
    ck = float(c.k[0][:, :, :16].float().abs().sum())
    cv = float(c.v[0][:, :, :16].float().abs().sum())
    print("pre-graph: curtok", cur.tolist(), "pos", positions.tolist(),
          "lens", c.seq_lens.tolist(), f"kck {ck:.3f} vck {cv:.3f}",
          flush=True)
    return orig(self, cur, positions, out_tokens, T, sp)
PipelineRunner._graph_decode = patched
torch.manual_seed(45)
ids = torch.randint(0, 1024, (2, 16))
sp = SamplingParams(temperature=0.8, top_p=0.9, max_new_tokens=12, seed=5)
for i in range(3):
    o = r.generate(ids, sp)
    print(f"o{i}:", o[0].tolist(), flush=True)
