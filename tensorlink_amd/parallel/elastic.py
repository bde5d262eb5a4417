"""Elastic recovery: re-form the process group after a rank loss.

The reference detects dead workers with heartbeat pings and drops them
from the distributed graph but leaves re-forming to operator restart
(``nodes/job_monitor.py`` health loop; SURVEY.md §2.6 "failure
detection"). Here the survivors recover THEMSELVES: a collective that
fails (peer socket closed / RCCL error) triggers a survivor rendezvous on
a side TCPStore, the group re-initializes with the survivor count, the
pipeline plan is recomputed for the smaller world and the model is
rebuilt from the same seeded init or checkpoint — so serving resumes
without outside intervention (Watchdog handles same-process restarts;
this handles lost peers).

Assumption: rank 0 survives (it hosts the rendezvous store, mirroring
the reference where the validator/coordinator is the fixed root).
"""

from __future__ import annotations

import datetime
import os
import time
from typing import Optional

import torch
import torch.distributed as dist

from tensorlink_amd.parallel.pipeline import PipelineRunner, SamplingParams
from tensorlink_amd.parallel.planner import plan_for_world


class ElasticRunner:
    """Pipeline runner wrapper that survives peer loss.

    ``generate`` retries once after a recovery: the failed request is
    re-run on the re-formed (smaller) group from scratch — generation
    state is per-request so nothing else is lost.
    """

    def __init__(self, model: str, device=None, seed: int = 0,
                 ckpt_dir: Optional[str] = None,
                 rendezvous_port: int = 29799,
                 grace_s: float = 3.0):
        self.model = model
        self.device = device or torch.device(
            "cuda" if torch.cuda.is_available() else "cpu")
        self.seed = seed
        # directory of per-stage safetensors (save_checkpoint writes it;
        # recovery re-assembles the new partitioning from it). Random
        # init is per-stage-seeded, so WITHOUT a checkpoint a re-formed
        # group serves a re-initialized model.
        self.ckpt_dir = ckpt_dir
        self.rendezvous_port = rendezvous_port
        self.grace_s = grace_s
        self.generation = 0              # bumps at every re-formation
        self.alive = True                # False once this rank is retired
        self._build()

    # ------------------------------------------------------------------
    def _build(self):
        rank = dist.get_rank() if dist.is_initialized() else 0
        world = dist.get_world_size() if dist.is_initialized() else 1
        plan = plan_for_world(self.model, world)
        self.runner = PipelineRunner(plan, rank, world, device=self.device,
                                     seed=self.seed)
        self.rank, self.world = rank, world
        if self.ckpt_dir and os.path.isdir(self.ckpt_dir):
            from tensorlink_amd.models.loader import \
                load_stage_from_stage_ckpt
            load_stage_from_stage_ckpt(self.runner.stage, self.ckpt_dir)

    # ------------------------------------------------------------------
    def generate(self, input_ids: torch.Tensor,
                 sampling: Optional[SamplingParams] = None):
        """Generate; on a peer failure re-form the group and retry once.
        Returns None on ranks that no longer host the pipeline."""
        if not self.alive:
            return None
        try:
            return self.runner.generate(input_ids, sampling)
        except RuntimeError:
            self.recover()
            if not self.alive:
                return None
            return self.runner.generate(input_ids, sampling)

    # ------------------------------------------------------------------
    def recover(self):
        """Survivor rendezvous + group re-formation + model rebuild."""
        old_rank = self.rank
        old_world = self.world
        if dist.is_initialized():
            dist.destroy_process_group()
        self.generation += 1
        # fresh port per generation so stale sockets never collide
        port = self.rendezvous_port + self.generation
        store = dist.TCPStore("127.0.0.1", port, is_master=(old_rank == 0),
                              timeout=datetime.timedelta(
                                  seconds=max(self.grace_s * 3, 10.0)),
                              wait_for_workers=False)
        store.set(f"alive_{old_rank}", "1")
        # give every survivor time to register before the roll call
        time.sleep(self.grace_s)
        survivors = []
        for r in range(old_world):
            try:
                probe = dist.TCPStore(
                    "127.0.0.1", port, is_master=False,
                    timeout=datetime.timedelta(seconds=1.0),
                    wait_for_workers=False)
                probe.get(f"alive_{r}")
                survivors.append(r)
            except Exception:
                pass
        new_world = len(survivors)
        new_rank = survivors.index(old_rank)
        backend = "nccl" if self.device.type == "cuda" else "gloo"
        dist.init_process_group(
            backend=backend, store=dist.PrefixStore(
                f"gen{self.generation}", store),
            rank=new_rank, world_size=new_world,
            timeout=datetime.timedelta(seconds=600))
        os.environ["RANK"] = str(new_rank)
        os.environ["WORLD_SIZE"] = str(new_world)
        self._build()

    def save_checkpoint(self, out_dir: Optional[str] = None):
        """Every rank dumps its stage shard (+ layer-range sidecar)."""
        from tensorlink_amd.models.loader import save_stage_to_safetensors
        out_dir = out_dir or self.ckpt_dir
        save_stage_to_safetensors(self.runner.stage, out_dir, self.rank)
        self.ckpt_dir = out_dir

    def shutdown(self):
        self.alive = False
        if dist.is_initialized():
            dist.destroy_process_group()
