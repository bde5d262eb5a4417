"""Node-state persistence (Keeper-lite).

The reference's ``Keeper`` periodically snapshots DHT/node state to
``logs/dht_state.json`` and reloads it with freshness filters
(``nodes/keeper.py:616-700``). Here the persisted state is the serving
history: metrics rollups, per-model demand and job table, written every
``interval_s`` and restored at engine start.
"""

from __future__ import annotations

import json
import os
import threading
import time
from typing import Optional

STATE_FILE = "logs/engine_state.json"
MAX_AGE_S = 30 * 24 * 3600      # reference: 30-day entity filter


class StateKeeper:
    def __init__(self, engine, path: str = STATE_FILE,
                 interval_s: float = 300.0):
        self.engine = engine
        self.path = path
        self.interval_s = interval_s
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None

    # ---- write ----
    def write_state(self):
        snap = {
            "t": time.time(),
            "metrics": self.engine.metrics.snapshot(),
            "demand": dict(self.engine.demand),
            "models": [j["model"] for j in self.engine.models()],
        }
        os.makedirs(os.path.dirname(self.path) or ".", exist_ok=True)
        tmp = self.path + ".tmp"
        with open(tmp, "w") as f:
            json.dump(snap, f)
        os.replace(tmp, self.path)
        return snap

    # ---- load ----
    def load_previous_state(self) -> Optional[dict]:
        if not os.path.exists(self.path):
            return None
        try:
            with open(self.path) as f:
                snap = json.load(f)
        except Exception:
            return None
        if time.time() - snap.get("t", 0) > MAX_AGE_S:
            return None
        for name, count in snap.get("demand", {}).items():
            self.engine.demand[name] += count
        return snap

    def start(self):
        self._thread = threading.Thread(target=self._loop, daemon=True)
        self._thread.start()
        return self

    def stop(self):
        self._stop.set()
        self.write_state()

    def _loop(self):
        while not self._stop.wait(self.interval_s):
            try:
                self.write_state()
            except Exception:
                pass
