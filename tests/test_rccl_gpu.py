"""RCCL-on-hardware coverage within a 1-GPU lease:
two ranks share cuda:0 with backend "nccl" (RCCL on ROCm), proving the
NCCL-backend code path — dtype/tag agreement, P2P send/recv ordering,
broadcast_obj — and pp2 generate/train equality vs single-rank, off
gloo. The 1→8 xGMI scaling curve itself is the driver's round-end job
(SCALE record); this is the closest a single-GPU lease can get.
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


def _rccl_gen_worker(rank, world, port, q):
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
        torch.cuda.set_device(0)
        init_distributed(backend="nccl")
        dev = torch.device("cuda", 0)
        # collective smoke: both ranks on one device over RCCL
        t = torch.ones(1024, device=dev) * (rank + 1)
        dist.all_reduce(t)
        assert float(t[0]) == 3.0
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


def test_pp2_generate_over_rccl_single_gpu():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = _port(1)
    ps = [ctx.Process(target=_rccl_gen_worker, args=(r, 2, port, q))
          for r in range(2)]
    for p in ps:
        p.start()
    try:
        status, out = q.get(timeout=180)
    finally:
        for p in ps:
            p.join(60)
            if p.is_alive():
                p.terminate()
    assert status == "ok", out

    # single-rank reference on the same seed/weights
    from tensorlink_amd.parallel.planner import plan_for_world
    from tensorlink_amd.parallel.pipeline import PipelineRunner, SamplingParams
    r1 = PipelineRunner(plan_for_world("tiny", 1), 0, 1,
                        device=torch.device("cuda", 0),
                        dtype=torch.bfloat16, seed=10)
    torch.manual_seed(7)
    ids = torch.randint(0, 1024, (4, 12))
    ref = r1.generate(ids, SamplingParams(max_new_tokens=6))
    assert torch.equal(out, ref.cpu())
