#!/usr/bin/env python3
"""tlctl — operator CLI (reference bin/run_node.py analog).

  tlctl serve  [--model NAME ...] [--port P] [--gpus N]   start the REST
      serving engine (SPMD under torchrun for N>1)
  tlctl plan   --model NAME [--gpus N] [--training]       print stage plan
  tlctl bench  [bench.py args...]                          run the benchmark
  tlctl health                                             device health
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))


def cmd_serve(args):
    from tensorlink_amd.config import EngineConfig
    from tensorlink_amd.engine.engine import InferenceEngine
    from tensorlink_amd.parallel.comm import init_distributed
    from tensorlink_amd.utils.logging import enable_file_logging, get_logger
    from tensorlink_amd.utils.state import StateKeeper
    from tensorlink_amd.utils.watchdog import Watchdog

    cfg = EngineConfig.load(args.config)
    log = get_logger("tensorlink_amd.serve")
    enable_file_logging()                 # logs/runtime.log, daily x 7
    rank, world = init_distributed()
    engine = InferenceEngine(rank=rank, world=world,
                             default_init=cfg.ml.init)
    if rank != 0:
        engine.worker_loop()
        return

    models = args.model or cfg.ml.default_models
    for name in models:
        log.info("loading %s ...", name)
        engine.load_model(
            name, continuous=args.continuous or cfg.ml.continuous,
            max_slots=args.max_slots, max_ctx=args.max_ctx,
            prefill_chunk=args.prefill_chunk or cfg.ml.prefill_chunk,
            prefix_caching=args.prefix_caching or cfg.ml.prefix_caching,
            speculative=args.speculative)
    keeper = StateKeeper(engine)
    keeper.load_previous_state()
    keeper.start()
    Watchdog(engine, job_ttl_s=cfg.ml.job_ttl_s).start()

    from tensorlink_amd.api.server import TensorlinkAPI
    api = TensorlinkAPI(engine, host=cfg.node.endpoint_host,
                        port=args.port or cfg.node.endpoint_port)
    print(f"serving on http://{api.host}:{api.port}  "
          f"(models: {', '.join(models)})")
    try:
        api.start(background=False)
    finally:
        keeper.stop()
        engine.shutdown()


def cmd_plan(args):
    from tensorlink_amd.parallel.planner import ModelParser
    parser = ModelParser(n_workers=args.gpus)
    plan = parser.create_distributed_config(
        args.model, batch_size=args.batch, seq_len=args.seq_len,
        training=args.training,
        num_stages=args.gpus if args.pp else None)
    print(plan.describe())


def cmd_bench(args, extra):
    os.execv(sys.executable, [sys.executable,
                              os.path.join(os.path.dirname(__file__),
                                           "bench.py")] + extra)


def cmd_export(args):
    import torch

    from tensorlink_amd.models import build_full_model, get_config
    from tensorlink_amd.models.loader import (init_random_stage,
                                              load_stage_from_checkpoint,
                                              save_hf_checkpoint)
    cfg = get_config(args.model)
    m = build_full_model(cfg)
    if args.ckpt_dir:
        n = load_stage_from_checkpoint(m, args.ckpt_dir,
                                       dtype=torch.bfloat16)
        print(f"loaded {n} tensors from {args.ckpt_dir}")
    else:
        init_random_stage(m, dtype=torch.bfloat16, seed=args.seed)
        print(f"random init (seed {args.seed})")
    out = save_hf_checkpoint(m, args.out)
    print(f"exported {cfg.name} -> {out} (HF safetensors layout)")


def cmd_status(args):
    if args.url:
        import json
        import urllib.request
        for ep in ("/stats", "/models", "/node-info"):
            with urllib.request.urlopen(args.url.rstrip("/") + ep,
                                        timeout=10) as r:
                print(ep, json.dumps(json.load(r), indent=2))
        return
    import torch

    from tensorlink_amd.engine.engine import InferenceEngine
    from tensorlink_amd.utils.dashboard import render_status
    eng = InferenceEngine(rank=0, world=1)
    print(render_status(eng))


def cmd_health(args):
    import json
    from tensorlink_amd.utils.watchdog import check_gpu_health
    print(json.dumps(check_gpu_health(), indent=2))


def main():
    p = argparse.ArgumentParser(prog="tlctl")
    sub = p.add_subparsers(dest="cmd", required=True)

    s = sub.add_parser("serve")
    s.add_argument("--model", action="append")
    s.add_argument("--port", type=int)
    s.add_argument("--config")
    s.add_argument("--continuous", action="store_true",
                   help="continuous batching (slot scheduler; PP-aware "
                        "when world > 1)")
    s.add_argument("--max-slots", type=int, default=16)
    s.add_argument("--max-ctx", type=int, default=4096)
    s.add_argument("--prefill-chunk", type=int, default=None)
    s.add_argument("--speculative", action="store_true",
                   help="prompt-lookup speculation when a single "
                        "request is decoding (exact greedy)")
    s.add_argument("--prefix-caching", action="store_true",
                   help="reuse KV pages across requests sharing a "
                        "prompt prefix")

    pl = sub.add_parser("plan")
    pl.add_argument("--model", required=True)
    pl.add_argument("--gpus", type=int, default=8)
    pl.add_argument("--batch", type=int, default=8)
    pl.add_argument("--seq-len", type=int, default=4096)
    pl.add_argument("--training", action="store_true")
    pl.add_argument("--pp", action="store_true",
                    help="force PP = --gpus instead of smallest fit")

    sub.add_parser("bench")
    sub.add_parser("health")
    st = sub.add_parser("status", help="terminal status dashboard "
                        "(VRAM bars, jobs, throughput)")
    st.add_argument("--url", default=None,
                    help="query a running server instead of local state")
    ex = sub.add_parser("export", help="export a model to the HF "
                        "safetensors layout (loadable by HF tooling)")
    ex.add_argument("--model", required=True,
                    help="preset name or checkpoint dir")
    ex.add_argument("--out", required=True)
    ex.add_argument("--ckpt-dir", default=None,
                    help="load weights from this HF-layout dir first "
                         "(default: seeded random init)")
    ex.add_argument("--seed", type=int, default=0)

    args, extra = p.parse_known_args()
    if args.cmd == "serve":
        cmd_serve(args)
    elif args.cmd == "plan":
        cmd_plan(args)
    elif args.cmd == "bench":
        cmd_bench(args, extra)
    elif args.cmd == "health":
        cmd_health(args)
    elif args.cmd == "status":
        cmd_status(args)
    elif args.cmd == "export":
        cmd_export(args)


if __name__ == "__main__":
    main()
