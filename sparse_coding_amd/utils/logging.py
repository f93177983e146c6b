"""Run logging: JSONL on disk, wandb if installed and requested.

The reference hard-depends on wandb + a secrets.json at import
(big_sweep.py:310-319); here logging is pluggable: a local JSONL metrics log
always works, and wandb is used only when available AND cfg.use_wandb.
"""

from __future__ import annotations

import json
import os
import time
from typing import Any, Dict, Optional


class RunLogger:
    def __init__(self, output_folder: str, name: str = "run", use_wandb: bool = False, config: Optional[Dict] = None):
        os.makedirs(output_folder, exist_ok=True)
        self.path = os.path.join(output_folder, f"{name}_metrics.jsonl")
        self._f = open(self.path, "a", buffering=1)
        self._step = 0
        self.wandb_run = None
        if use_wandb:
            try:
                import wandb

                self.wandb_run = wandb.init(project="sparse_coding_amd", name=name, config=config or {})
            except Exception as e:  # noqa: BLE001
                print(f"[logger] wandb unavailable ({e}); logging to {self.path} only")

    def log(self, metrics: Dict[str, Any], commit: bool = True) -> None:
        rec = {"_step": self._step, "_time": time.time()}
        for k, v in metrics.items():
            if hasattr(v, "item"):
                try:
                    v = v.item()
                except Exception:  # noqa: BLE001
                    continue
            if isinstance(v, (int, float, str, bool)) or v is None:
                rec[k] = v
        self._f.write(json.dumps(rec) + "\n")
        if self.wandb_run is not None:
            self.wandb_run.log(metrics, commit=commit)
        if commit:
            self._step += 1

    # picklable across mp.Process boundaries (cluster dispatch): drop the
    # open file / wandb handle; children reopen the JSONL in append mode.
    def __getstate__(self):
        state = self.__dict__.copy()
        state["_f"] = None
        state["wandb_run"] = None
        return state

    def __setstate__(self, state):
        self.__dict__.update(state)
        self._f = open(self.path, "a", buffering=1)

    def log_image(self, key: str, fig) -> None:
        folder = os.path.dirname(self.path)
        img_dir = os.path.join(folder, "images")
        os.makedirs(img_dir, exist_ok=True)
        fig.savefig(os.path.join(img_dir, f"{key.replace('/', '_')}_{self._step}.png"))
        if self.wandb_run is not None:
            import wandb

            self.wandb_run.log({key: wandb.Image(fig)}, commit=False)

    def close(self) -> None:
        self._f.close()
        if self.wandb_run is not None:
            self.wandb_run.finish()
