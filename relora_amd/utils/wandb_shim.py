"""wandb facade: real wandb if importable, else an offline no-op recorder.

The reference makes wandb mandatory on rank 0 (reference torchrun_main.py:404-412,
923-942). This image has no network and no wandb package, so `relora_amd`
routes all experiment tracking through this module: when wandb is present it
is used as-is; otherwise a no-op implementation records the last-logged
metrics in memory (handy for tests) and optionally appends JSONL to
`{save_dir}/wandb_offline.jsonl` when `init(dir=...)` is given.
"""

import json
import os
import types
import uuid

try:  # pragma: no cover - exercised only when wandb is installed
    import wandb as _real_wandb
except Exception:  # ImportError or anything broken in a partial install
    _real_wandb = None


class AlertLevel:
    INFO = "INFO"
    WARN = "WARN"
    ERROR = "ERROR"


class _Run:
    def __init__(self, project=None, name=None, id=None, dir=None, config=None, **kwargs):
        self.project = project
        self.name = name or f"run_{uuid.uuid4().hex[:8]}"
        self.id = id or uuid.uuid4().hex[:12]
        self.dir = dir
        self.config = _Config(config or {})
        self.history = []
        self._file = None
        if dir is not None:
            os.makedirs(dir, exist_ok=True)
            self._file = open(os.path.join(dir, "wandb_offline.jsonl"), "a")

    def save(self, *args, **kwargs):
        pass

    def finish(self):
        if self._file:
            self._file.close()
            self._file = None


class _Config(dict):
    def update(self, d, allow_val_change=True):
        if hasattr(d, "__dict__") and not isinstance(d, dict):
            d = vars(d)
        dict.update(self, d)

    def __getattr__(self, k):
        try:
            return self[k]
        except KeyError as e:
            raise AttributeError(k) from e

    def __setattr__(self, k, v):
        self[k] = v


class _NoopWandb(types.ModuleType):
    """Subset of the wandb API used by the trainer."""

    def __init__(self):
        super().__init__("wandb")
        self.run = None
        self.AlertLevel = AlertLevel

    @property
    def config(self):
        return self.run.config if self.run else _Config()

    def init(self, project=None, name=None, id=None, resume=None, dir=None,
             config=None, tags=None, notes=None, **kwargs):
        self.run = _Run(project=project, name=name, id=id, dir=dir, config=config)
        return self.run

    def log(self, metrics, step=None):
        if self.run is None:
            return
        rec = dict(metrics)
        if step is not None:
            rec["_step"] = step
        self.run.history.append(rec)
        if self.run._file:
            try:
                self.run._file.write(json.dumps(rec, default=str) + "\n")
                self.run._file.flush()
            except Exception:
                pass

    def watch(self, *args, **kwargs):
        pass

    def alert(self, title=None, text=None, level=None, **kwargs):
        pass

    def save(self, *args, **kwargs):
        pass

    def finish(self):
        if self.run is not None:
            self.run.finish()
            self.run = None


wandb = _real_wandb if _real_wandb is not None else _NoopWandb()
