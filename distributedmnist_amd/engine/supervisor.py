"""Checkpoint supervisor — replaces tf.train.Supervisor/Saver
(distributed_train.py:244-262,405-408; layout per SURVEY.md section 5.4).

Layout kept compatible with the reference's consumer (nn_eval.py:70-88):
  train_dir/model.ckpt-<step>     torch.save payload
  train_dir/checkpoint            index file, TF text format:
      model_checkpoint_path: "model.ckpt-<step>"
      all_model_checkpoint_paths: "model.ckpt-<step>"
The evaluator parses the step from the filename suffix exactly as the
reference does.
"""

from __future__ import annotations

import os
import re
import time

import torch


class Supervisor:
    def __init__(self, train_dir: str, save_interval_secs: float = 20.0,
                 is_chief: bool = True, keep_last: int = 5):
        self.train_dir = train_dir
        self.save_interval_secs = save_interval_secs
        self.is_chief = is_chief
        self.keep_last = keep_last
        self._last_save = 0.0
        if is_chief:
            os.makedirs(train_dir, exist_ok=True)

    # -- save ----------------------------------------------------------
    def maybe_save(self, step: int, payload) -> bool:
        """`payload` may be a dict or a zero-arg callable returning one.
        Pass a callable from the train loop: the payload build is a full
        D2H copy of every buffer, so it must only happen when a save is
        actually due (chief + interval elapsed), not every step."""
        if not self.is_chief:
            return False
        now = time.time()
        if now - self._last_save < self.save_interval_secs:
            return False
        self.save(step, payload() if callable(payload) else payload)
        return True

    def save(self, step: int, payload):
        if not self.is_chief:
            return
        if callable(payload):
            payload = payload()
        name = f"model.ckpt-{step}"
        path = os.path.join(self.train_dir, name)
        tmp = path + ".tmp"
        torch.save(dict(payload, step=step), tmp)
        os.replace(tmp, path)
        with open(os.path.join(self.train_dir, "checkpoint"), "w") as f:
            f.write(f'model_checkpoint_path: "{name}"\n')
            f.write(f'all_model_checkpoint_paths: "{name}"\n')
        self._last_save = time.time()
        self._gc()

    def _gc(self):
        ckpts = sorted_checkpoints(self.train_dir)
        for _, path in ckpts[:-self.keep_last]:
            try:
                os.remove(path)
            except OSError:
                pass

    # -- restore -------------------------------------------------------
    @staticmethod
    def latest_checkpoint(train_dir: str):
        """(step, path) of the newest checkpoint or None.

        Reads the index file first (reference get_checkpoint_state), falls
        back to a glob."""
        idx = os.path.join(train_dir, "checkpoint")
        if os.path.exists(idx):
            with open(idx) as f:
                for line in f:
                    m = re.match(r'model_checkpoint_path: "(.*)"', line.strip())
                    if m:
                        path = os.path.join(train_dir, os.path.basename(m.group(1)))
                        if os.path.exists(path):
                            sm = re.search(r"-(\d+)$", path)
                            return (int(sm.group(1)) if sm else 0), path
        ck = sorted_checkpoints(train_dir)
        return ck[-1] if ck else None

    @staticmethod
    def restore(train_dir: str, map_location="cpu"):
        """Returns (step, payload) or None."""
        latest = Supervisor.latest_checkpoint(train_dir)
        if latest is None:
            return None
        step, path = latest
        # tensor-only payload (tensors + str/int containers) — loads under
        # weights_only=True, so a tampered checkpoint cannot execute code
        payload = torch.load(path, map_location=map_location, weights_only=True)
        return step, payload


def sorted_checkpoints(train_dir: str):
    out = []
    if not os.path.isdir(train_dir):
        return out
    for fn in os.listdir(train_dir):
        m = re.match(r"model\.ckpt-(\d+)$", fn)
        if m:
            out.append((int(m.group(1)), os.path.join(train_dir, fn)))
    return sorted(out)
