"""GPU smokes for the big BASELINE configs: Mixtral-8x7B with fp8 experts
and Llama-3-70B on a single 288 GB MI355X (PP=1)."""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from tensorlink_amd.parallel.planner import plan_for_world
from tensorlink_amd.parallel.pipeline import PipelineRunner, SamplingParams


def smoke(name, quantize=None, batch=2, prompt=64, new=16, vocab=32000):
    t0 = time.time()
    plan = plan_for_world(name, 1, batch_size=batch * 2, seq_len=prompt + new * 2)
    r = PipelineRunner(plan, 0, 1, device=torch.device("cuda:0"),
                       quantize=quantize)
    mem = torch.cuda.memory_allocated() / 2**30
    print(f"{name} (quant={quantize}): init {time.time()-t0:.0f}s, "
          f"weights {mem:.0f} GiB", flush=True)
    ids = torch.randint(0, vocab, (batch, prompt))
    out, st = r.generate(ids, SamplingParams(max_new_tokens=new),
                         return_stats=True)
    dec = st.get("decode_tokens_per_s") or 0
    print(f"  decode ok: out {tuple(out.shape)}, "
          f"{st['output_tokens_per_s']:.0f} tok/s total, "
          f"{dec:.0f} tok/s decode", flush=True)
    del r
    torch.cuda.empty_cache()


if __name__ == "__main__":
    which = sys.argv[1] if len(sys.argv) > 1 else "both"
    if which in ("both", "all", "mixtral"):
        smoke("mistralai/Mixtral-8x7B-v0.1", quantize="fp8", vocab=32000)
    if which in ("both", "all", "llama70b"):
        smoke("meta-llama/Llama-3-70B", batch=8, new=32, vocab=128256)
    if which in ("all", "qwen3moe"):
        smoke("Qwen/Qwen3-30B-A3B", vocab=151936)
    if which in ("all", "qwen7b-fp8"):
        smoke("Qwen/Qwen2.5-7B-Instruct", quantize="fp8-dense",
              batch=8, new=32, vocab=152064)
