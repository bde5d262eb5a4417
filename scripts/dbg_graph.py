import sys, os, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from tensorlink_amd.parallel.planner import plan_for_world
from tensorlink_amd.parallel.pipeline import PipelineRunner, SamplingParams

plan = plan_for_world("tiny", 1)
r = PipelineRunner(plan, 0, 1, device=torch.device("cuda:0"), dtype=torch.bfloat16)
torch.manual_seed(9)
ids = torch.randint(0, 1024, (4, 16))
out_graph = r.generate(ids, SamplingParams(max_new_tokens=12)).cpu()
print("no_graph flag after:", r._no_graph, "graph obj:", r._decode_graph is not None)
r2 = PipelineRunner(plan, 0, 1, device=torch.device("cuda:0"), dtype=torch.bfloat16)
out_eager = r2.generate(ids, SamplingParams(temperature=1e-6, top_k=1, max_new_tokens=12)).cpu()
print("graph:", out_graph.tolist())
print("eager:", out_eager.tolist())
# also eager greedy with TL_NO_GRAPH via env on SAME runner kind
os.environ["TL_NO_GRAPH"] = "1"
r3 = PipelineRunner(plan, 0, 1, device=torch.device("cuda:0"), dtype=torch.bfloat16)
out3 = r3.generate(ids, SamplingParams(max_new_tokens=12)).cpu()
print("eager-greedy:", out3.tolist())
print("graph==eager-greedy:", torch.equal(out_graph, out3))
print("eager-sample==eager-greedy:", torch.equal(out_eager, out3))

del os.environ["TL_NO_GRAPH"]
ids2 = torch.randint(0, 1024, (4, 16))
out2 = r.generate(ids2, SamplingParams(max_new_tokens=12)).cpu()
out2e = r2.generate(ids2, SamplingParams(temperature=1e-6, top_k=1, max_new_tokens=12)).cpu()
os.environ["TL_NO_GRAPH"] = "1"
out2g = r3.generate(ids2, SamplingParams(max_new_tokens=12)).cpu()
print("2nd: graph==eager-sample:", torch.equal(out2, out2e))
print("2nd: graph==eager-greedy:", torch.equal(out2, out2g))
print("g:", out2[1].tolist())
print("e:", out2e[1].tolist())
print("n:", out2g[1].tolist())
