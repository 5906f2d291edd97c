"""Experiment logging: wandb when importable, JSONL fallback otherwise
(reference logs exclusively to wandb, train_dalle.py:463-476; here the same
signals always land in ``<output>/log.jsonl`` so offline runs keep metrics).
"""

import json
import time
from pathlib import Path


class RunLogger:
    def __init__(self, project, config=None, enabled=True, use_wandb=True,
                 output_dir='.', run_name=None, entity=None):
        self.enabled = enabled
        self.wandb = None
        self._jsonl = None
        if not enabled:
            return
        if use_wandb:
            try:
                import wandb
                self.wandb = wandb
                wandb.init(project=project, name=run_name, entity=entity,
                           config=config or {}, resume=False)
            except Exception:
                self.wandb = None
        out = Path(output_dir)
        out.mkdir(parents=True, exist_ok=True)
        self._jsonl = open(out / 'log.jsonl', 'a')
        self.log({'event': 'run_start', 'project': project,
                  'config': _sanitize(config or {})})

    def log(self, metrics, step=None):
        if not self.enabled:
            return
        if self.wandb is not None:
            loggable = {k: v for k, v in metrics.items() if not isinstance(v, (dict, str))}
            if loggable:
                self.wandb.log(loggable, step=step)
        if self._jsonl is not None:
            rec = {'t': time.time(), **_sanitize(metrics)}
            if step is not None:
                rec['step'] = step
            self._jsonl.write(json.dumps(rec) + '\n')
            self._jsonl.flush()

    def log_image(self, tag, tensor, caption=None, step=None):
        if not self.enabled or self.wandb is None:
            return
        self.wandb.log({tag: self.wandb.Image(tensor, caption=caption)}, step=step)

    def save(self, path):
        if self.enabled and self.wandb is not None:
            self.wandb.save(str(path))

    def finish(self):
        if not self.enabled:
            return
        if self.wandb is not None:
            self.wandb.finish()
        if self._jsonl is not None:
            self._jsonl.close()

    def log_artifact(self, path, name='trained-dalle', type_='model'):
        """Model-artifact upload per save (reference train_dalle.py:584-587,
        wandb.save + artifact logging); JSONL mode records the path only."""
        if not self.enabled:
            return
        if self.wandb is not None:
            try:
                art = self.wandb.Artifact(name, type=type_)
                art.add_file(str(path))
                self.wandb.log_artifact(art)
            except Exception:
                pass
        else:
            self.log({'checkpoint_saved': str(path)})


def _sanitize(obj):
    if isinstance(obj, dict):
        return {k: _sanitize(v) for k, v in obj.items()}
    if isinstance(obj, (list, tuple)):
        return [_sanitize(v) for v in obj]
    if isinstance(obj, (int, float, str, bool)) or obj is None:
        return obj
    return str(obj)
