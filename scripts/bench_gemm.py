"""A/B the deterministic GEMM family (streaming M<=64 / tiled M>64) vs
hipBLASLt on the Qwen2.5-7B serving shapes. Run on a GPU box:

    python scripts/bench_gemm.py [--csv gpurun_out/gemm_ab.csv]
"""
import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from tensorlink_amd import ops  # noqa: E402


def timeit(fn, iters=50):
    for _ in range(10):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6  # us


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--csv", default=None)
    args = ap.parse_args()
    C = ops._require_ext()
    rows = []
    shapes = [
        # decode / serving-M shapes (Qwen2.5-7B)
        (256, 4608, 3584, "qkv"),
        (256, 3584, 3584, "o"),
        (256, 37888, 3584, "gate_up"),
        (256, 3584, 18944, "down"),
        (256, 151936, 3584, "lm_head"),
        (512, 4608, 3584, "qkv_m512"),
        (512, 37888, 3584, "gate_up_m512"),
        (512, 3584, 18944, "down_m512"),
        (128, 4608, 3584, "qkv_m128"),
        (96, 4608, 3584, "qkv_m96"),
        (64, 4608, 3584, "qkv_m64"),
        (16, 3584, 3584, "o_m16"),
        (1, 4608, 3584, "qkv_m1"),
    ]
    for M, N, K, name in shapes:
        x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
        w = torch.randn(N, K, device="cuda", dtype=torch.bfloat16) / K ** 0.5
        t_lib = timeit(lambda: torch.nn.functional.linear(x, w))
        t_tl = timeit(lambda: C.skinny_gemm(x, w, None))
        # correctness spot check
        ref = torch.nn.functional.linear(x.float(), w.float())
        got = C.skinny_gemm(x, w, None).float()
        err = (got - ref).abs().max().item()
        wb = N * K * 2 / 1e9
        line = (f"{name:14s} M{M:4d} N{N:6d} K{K:6d}: "
                f"lib {t_lib:7.1f}us ({wb / t_lib * 1e3:5.2f} TB/s)  "
                f"tl {t_tl:7.1f}us ({wb / t_tl * 1e3:5.2f} TB/s)  "
                f"maxerr {err:.3f}")
        print(line, flush=True)
        rows.append((name, M, N, K, t_lib, t_tl, err))
    if args.csv:
        os.makedirs(os.path.dirname(args.csv), exist_ok=True)
        with open(args.csv, "w") as f:
            f.write("name,M,N,K,lib_us,tl_us,maxerr\n")
            for r in rows:
                f.write(",".join(str(v) for v in r) + "\n")


if __name__ == "__main__":
    main()
