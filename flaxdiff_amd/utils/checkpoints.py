"""Checkpoint save/restore.

Logical schema mirrors the reference's Orbax pytree (simple_trainer.py:369-389,
SURVEY.md §5.4):
    {rngs, state: {params, ema_params, opt_state, step, rngs},
     best_state: {...}, best_loss, epoch}
Directory layout: <base>/<name>/<step>/ with torch.save payloads and a JSON
config manifest (the run config round-trips at inference — §5.6).
Writes are rank-0 only and happen on a background thread (the reference's
AsyncCheckpointer equivalent).
"""
from __future__ import annotations

import json
import shutil
import threading
from pathlib import Path
from typing import Any, Dict, Optional

import torch


class CheckpointManager:
    def __init__(self, base_dir: str, name: str, max_to_keep: int = 2):
        self.dir = Path(base_dir) / name
        self.dir.mkdir(parents=True, exist_ok=True)
        self.max_to_keep = max_to_keep
        self._thread: Optional[threading.Thread] = None

    # ------------------------------------------------------------------
    def _step_dirs(self):
        out = []
        for d in self.dir.iterdir() if self.dir.exists() else []:
            if d.is_dir() and d.name.isdigit():
                out.append(int(d.name))
        return sorted(out)

    def latest_step(self) -> Optional[int]:
        steps = self._step_dirs()
        return steps[-1] if steps else None

    # ------------------------------------------------------------------
    def save(self, step: int, payload: Dict[str, Any], config: Optional[dict] = None,
             block: bool = False):
        """Async save; payload tensors are cloned to CPU synchronously first."""
        cpu_payload = _to_cpu(payload)
        self.wait()

        def _write():
            d = self.dir / str(step)
            tmp = self.dir / f".tmp_{step}"
            if tmp.exists():
                shutil.rmtree(tmp)
            tmp.mkdir(parents=True)
            torch.save(cpu_payload, tmp / "state.pt")
            if config is not None:
                with open(tmp / "config.json", "w") as f:
                    json.dump(config, f, indent=2, default=str)
            if d.exists():
                shutil.rmtree(d)
            tmp.rename(d)
            self._gc()

        self._thread = threading.Thread(target=_write, daemon=True)
        self._thread.start()
        if block:
            self.wait()

    def wait(self):
        if self._thread is not None and self._thread.is_alive():
            self._thread.join()
        self._thread = None

    def _gc(self):
        steps = self._step_dirs()
        while len(steps) > self.max_to_keep:
            victim = steps.pop(0)
            shutil.rmtree(self.dir / str(victim), ignore_errors=True)

    # ------------------------------------------------------------------
    def load(self, step: Optional[int] = None) -> Optional[Dict[str, Any]]:
        if step is None:
            step = self.latest_step()
        if step is None:
            return None
        path = self.dir / str(step) / "state.pt"
        if not path.exists():
            return None
        payload = torch.load(path, map_location="cpu", weights_only=False)
        payload["_step_dir"] = str(step)
        return payload

    def load_config(self, step: Optional[int] = None) -> Optional[dict]:
        if step is None:
            step = self.latest_step()
        if step is None:
            return None
        path = self.dir / str(step) / "config.json"
        if not path.exists():
            return None
        with open(path) as f:
            return json.load(f)


def _to_cpu(obj):
    if torch.is_tensor(obj):
        return obj.detach().to("cpu", copy=True)
    if isinstance(obj, dict):
        return {k: _to_cpu(v) for k, v in obj.items()}
    if isinstance(obj, (list, tuple)):
        t = [_to_cpu(v) for v in obj]
        return type(obj)(t) if not isinstance(obj, tuple) else tuple(t)
    return obj
