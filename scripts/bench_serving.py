"""Serving load benchmark: Poisson request arrivals against the
continuous batcher; reports throughput and TTFT/TPOT percentiles.

The reference ships no load-testing tool (it publishes no
performance numbers at all); this is the vLLM-``benchmark_serving``-style
harness for this engine. Runs the batcher directly (no HTTP overhead) —
point it at a GPU box for real numbers, or at the CPU tier for a smoke:

    python scripts/bench_serving.py --model tiny --rate 50 \
        --num-requests 20 --prompt-len 64 --new-tokens 16

    # MI355X, chunked prefill + prefix caching:
    python scripts/bench_serving.py --model Qwen/Qwen2.5-7B-Instruct \
        --rate 8 --num-requests 128 --prompt-len 1024 --new-tokens 128 \
        --prefill-chunk 512 --prefix-caching --shared-prefix 512
"""

from __future__ import annotations

import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import json
import random
import threading
import time

import torch

from tensorlink_amd.engine.batcher import ContinuousBatcher
from tensorlink_amd.parallel.pipeline import PipelineRunner
from tensorlink_amd.parallel.planner import plan_for_world


def pct(vals, p):
    if not vals:
        return None
    vals = sorted(vals)
    return vals[min(len(vals) - 1, int(p / 100 * len(vals)))]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="tiny")
    ap.add_argument("--rate", type=float, default=10.0,
                    help="mean request arrival rate (req/s, Poisson)")
    ap.add_argument("--num-requests", type=int, default=32)
    ap.add_argument("--prompt-len", type=int, default=128)
    ap.add_argument("--new-tokens", type=int, default=32)
    ap.add_argument("--max-slots", type=int, default=16)
    ap.add_argument("--max-ctx", type=int, default=2048)
    ap.add_argument("--prefill-chunk", type=int, default=None)
    ap.add_argument("--prefix-caching", action="store_true")
    ap.add_argument("--speculative", action="store_true")
    ap.add_argument("--shared-prefix", type=int, default=0,
                    help="tokens of prompt shared by ALL requests "
                         "(exercises prefix caching)")
    ap.add_argument("--seed", type=int, default=0)
    args = ap.parse_args()

    device = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    runner = PipelineRunner(plan_for_world(args.model, 1), 0, 1,
                            device=device)
    batcher = ContinuousBatcher(
        runner, max_slots=args.max_slots, max_ctx=args.max_ctx,
        prefill_chunk=args.prefill_chunk,
        prefix_caching=args.prefix_caching,
        speculative=args.speculative).start()

    rng = random.Random(args.seed)
    g = torch.Generator().manual_seed(args.seed)
    vocab = runner.config.vocab_size
    shared = torch.randint(0, vocab, (args.shared_prefix,), generator=g)

    results = []
    lock = threading.Lock()

    def client(i, prompt):
        t0 = time.perf_counter()
        req = batcher.submit(prompt, max_new_tokens=args.new_tokens)
        t_first = None
        n = 0
        for _ in req.stream(timeout=600):
            if t_first is None:
                t_first = time.perf_counter()
            n += 1
        t1 = time.perf_counter()
        with lock:
            results.append({"ttft": t_first - t0 if t_first else None,
                            "latency": t1 - t0, "tokens": n})

    threads = []
    t_start = time.perf_counter()
    for i in range(args.num_requests):
        tail = torch.randint(0, vocab,
                             (max(1, args.prompt_len - args.shared_prefix),),
                             generator=g)
        prompt = torch.cat([shared, tail])
        th = threading.Thread(target=client, args=(i, prompt))
        th.start()
        threads.append(th)
        time.sleep(rng.expovariate(args.rate))
    for th in threads:
        th.join(timeout=900)
    elapsed = time.perf_counter() - t_start
    batcher.stop()

    ttfts = [r["ttft"] for r in results if r["ttft"] is not None]
    tpots = [(r["latency"] - r["ttft"]) / max(1, r["tokens"] - 1)
             for r in results if r["ttft"] is not None and r["tokens"] > 1]
    total_tokens = sum(r["tokens"] for r in results)
    print(json.dumps({
        "model": args.model, "num_requests": len(results),
        "rate_req_s": args.rate, "elapsed_s": elapsed,
        "output_tokens_per_s": total_tokens / elapsed,
        "requests_per_s": len(results) / elapsed,
        "ttft_p50_s": pct(ttfts, 50), "ttft_p95_s": pct(ttfts, 95),
        "tpot_p50_s": pct(tpots, 50), "tpot_p95_s": pct(tpots, 95),
        "latency_p50_s": pct([r["latency"] for r in results], 50),
        "prefix_cache_hit_tokens": getattr(batcher.cache, "hits", 0),
        "spec_accepted_tokens": getattr(batcher, "spec_accepted", 0),
        "preemptions": getattr(batcher, "preemptions", 0),
        "scheduler_steps": batcher.steps,
        "config": {"prompt_len": args.prompt_len,
                   "new_tokens": args.new_tokens,
                   "max_slots": args.max_slots,
                   "prefill_chunk": args.prefill_chunk,
                   "prefix_caching": args.prefix_caching,
                   "shared_prefix": args.shared_prefix}}))


if __name__ == "__main__":
    main()
