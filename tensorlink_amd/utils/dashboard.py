"""Terminal status dashboard (reference ``p2p/torch_node.py:963-1049``:
``print_ui_status`` draws VRAM/RAM bars, peer/job tables to the console).
Here the "peers" are GPU ranks on one node; ``render_status`` returns the
text block so tlctl/status endpoints and tests can reuse it.
"""

from __future__ import annotations

from typing import List

import torch


def _bar(frac: float, width: int = 30) -> str:
    frac = max(0.0, min(1.0, frac))
    fill = int(round(frac * width))
    return "[" + "#" * fill + "-" * (width - fill) + f"] {frac * 100:5.1f}%"


def gpu_lines() -> List[str]:
    lines = []
    if torch.cuda.is_available():
        for d in range(torch.cuda.device_count()):
            free, total = torch.cuda.mem_get_info(d)
            used = total - free
            name = torch.cuda.get_device_name(d)
            lines.append(f"GPU{d} {name}: "
                         f"{_bar(used / total)} "
                         f"{used / 2**30:.1f}/{total / 2**30:.0f} GiB")
    else:
        import psutil
        vm = psutil.virtual_memory()
        lines.append(f"CPU RAM: {_bar(vm.percent / 100)} "
                     f"{vm.used / 2**30:.1f}/{vm.total / 2**30:.0f} GiB")
    return lines


def render_status(engine) -> str:
    """Text dashboard for an InferenceEngine (VRAM bars, job table,
    throughput counters — the reference's UI status block)."""
    out = ["== tensorlink-amd status =="]
    out += gpu_lines()
    out.append(f"ranks: {engine.world}  models loaded: {len(engine.jobs)}")
    if engine.jobs:
        out.append(f"{'model':<40} {'state':<8} {'reqs':>6} {'tokens':>10} "
                   f"{'batcher':>8}")
        for name, job in engine.jobs.items():
            out.append(f"{name:<40} {job.state:<8} "
                       f"{job.requests_served:>6} "
                       f"{job.tokens_generated:>10} "
                       f"{'on' if job.batcher is not None else 'off':>8}")
    m = engine.metrics.snapshot() if hasattr(engine.metrics,
                                             "snapshot") else {}
    if m:
        out.append(f"uptime: {m.get('uptime_s', 0):.0f}s  "
                   f"requests: {m.get('requests_total', 0)}  "
                   f"errors: {m.get('errors_total', 0)}  "
                   f"tokens: {m.get('tokens_total', 0)}")
    demand = engine.model_demand()
    if demand:
        top = sorted(demand.items(), key=lambda kv: -kv[1])[:5]
        out.append("demand: " + ", ".join(f"{k}={v}" for k, v in top))
    return "\n".join(out)
