"""RCCL-on-hardware coverage within a 1-GPU lease.

RCCL refuses two ranks on one device ("Duplicate GPU detected", verified
on MI355X — the communicator requires distinct devices), so a single
lease cannot run a real 2-rank RCCL pipeline. What CAN run, and does
here:

1. a single-rank RCCL process group executing real collectives on
   device memory (backend init, dtype plumbing, RCCL kernels on xGMI-
   attached HBM);
2. the full 2-process SPMD pipeline with BOTH ranks computing on the
   one GPU, transported over gloo with host staging — every pipeline
   send/recv path runs against real GPU activations, and the result
   must equal single-rank generation bitwise.

The 1→8 xGMI scaling itself is the driver's round-end job (SCALE).
"""

import os
import sys

import pytest
import torch
import torch.multiprocessing as mp

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _port(offset: int) -> int:
    return 21000 + (os.getpid() * 17 + offset * 131) % 20000


def test_rccl_single_rank_collectives():
    """RCCL (nccl backend) init + collectives on this GPU."""
    import torch.distributed as dist
    os.environ.update(RANK="0", WORLD_SIZE="1", LOCAL_RANK="0",
                      MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(_port(3)))
    dist.init_process_group("nccl", rank=0, world_size=1)
    try:
        dev = torch.device("cuda", 0)
        t = torch.ones(1 << 20, device=dev, dtype=torch.bfloat16)
        dist.all_reduce(t)
        assert float(t[0]) == 1.0
        dist.broadcast(t, 0)
        x = torch.randn(8, 1024, device=dev)
        dist.all_gather([torch.empty_like(x)], x)
        dist.barrier()
    finally:
        dist.destroy_process_group()
    for k in ("MASTER_ADDR", "MASTER_PORT", "RANK", "WORLD_SIZE"):
        os.environ.pop(k, None)


def _gpu_pp2_worker(rank, world, port, q):
    sys.path.insert(0, REPO)
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK="0",  # both ranks share the one GPU
                      MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port))
    import torch
    import torch.distributed as dist
    from tensorlink_amd.parallel.comm import init_distributed
    from tensorlink_amd.parallel.planner import plan_for_world
    from tensorlink_amd.parallel.pipeline import PipelineRunner, SamplingParams
    try:
        init_distributed(backend="gloo")
        dev = torch.device("cuda", 0)
        plan = plan_for_world("tiny", world)
        r = PipelineRunner(plan, rank, world, device=dev,
                           dtype=torch.bfloat16, seed=10)
        torch.manual_seed(7)
        ids = torch.randint(0, 1024, (4, 12)) if rank == 0 else None
        out = r.generate(ids, SamplingParams(max_new_tokens=6))
        if rank == 0:
            q.put(("ok", out.cpu()))
        dist.destroy_process_group()
    except Exception as e:  # surface the failure to the parent
        if rank == 0:
            q.put(("err", repr(e)))
        raise


def test_pp2_gpu_compute_host_transport():
    """Two pipeline ranks share the GPU (host-staged transport): output
    equals single-rank generation bitwise (deterministic GEMM family +
    per-row decode split make stages batch/placement independent)."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = _port(1)
    ps = [ctx.Process(target=_gpu_pp2_worker, args=(r, 2, port, q))
          for r in range(2)]
    for p in ps:
        p.start()
    try:
        status, out = q.get(timeout=300)
    finally:
        for p in ps:
            p.join(60)
            if p.is_alive():
                p.terminate()
    assert status == "ok", out

    # single-process reference: the same two per-stage inits chained
    # in one process on the same GPU (cf. tests/test_pipeline_multiproc)
    from tensorlink_amd.models.dense import build_stage
    from tensorlink_amd.models.loader import init_random_stage
    from tensorlink_amd.parallel.planner import plan_for_world
    dev = torch.device("cuda", 0)
    plan = plan_for_world("tiny", 2)
    stages = []
    for rk in range(2):
        st = build_stage(plan.config, plan.stage_for_rank(rk))
        init_random_stage(st, device=dev, dtype=torch.bfloat16,
                          seed=10 + rk)
        st.eval()
        stages.append(st)
    torch.manual_seed(7)
    ids = torch.randint(0, 1024, (4, 12)).to(dev)
    caches = [st.make_kv_cache(4, 32, dev, torch.bfloat16)
              for st in stages]
    toks = []
    with torch.no_grad():
        pos = torch.arange(12, device=dev, dtype=torch.int32) \
            .unsqueeze(0).expand(4, -1).contiguous()
        h = stages[0](ids, pos, kv_cache=caches[0], return_logits=False)
        logits = stages[1](h, pos, kv_cache=caches[1])
        tok = logits[:, -1].argmax(-1)
        toks.append(tok)
        for t in range(5):
            posd = torch.full((4, 1), 12 + t, device=dev,
                              dtype=torch.int32)
            h = stages[0](tok.unsqueeze(1), posd, kv_cache=caches[0],
                          return_logits=False)
            lg = stages[1](h, posd, kv_cache=caches[1])
            tok = lg.squeeze(1).argmax(-1)
            toks.append(tok)
    ref = torch.stack(toks, 1).cpu()
    assert torch.equal(out, ref)
