"""Scope-isolated summaries — "TensorBoard is AdaNet's UI" re-done without TF.

The reference isolates each candidate's charts by monkey-patching every
tf.summary entry point while user code runs and writing per-scope event
files (reference adanet/core/summary.py:42-973). Here each scope gets its
own JSONL event stream under <logdir>/<scope>/events.jsonl: the same metric
name from different candidates lands in files with identical relative names,
so charts overlay when loaded (the reference's design goal,
summary.py:262-296). No TensorBoard dependency (not installed on ROCm
boxes); the JSONL is trivially loadable with pandas/jq.
"""

from __future__ import annotations

import abc
import json
import os
import threading
import time
from typing import Dict, Optional

import torch


class Summary(abc.ABC):
    """Simple summary API: scalar/histogram/image/audio/text.

    Reference: adanet/core/summary.py:42-210 (the `Summary` ABC).
    """

    @abc.abstractmethod
    def scalar(self, name: str, tensor, family: Optional[str] = None):
        ...

    @abc.abstractmethod
    def histogram(self, name: str, tensor, family: Optional[str] = None):
        ...

    @abc.abstractmethod
    def image(self, name: str, tensor, family: Optional[str] = None):
        ...

    @abc.abstractmethod
    def audio(self, name: str, tensor, sample_rate, family: Optional[str] = None):
        ...


def _to_scalar(value):
    if isinstance(value, torch.Tensor):
        return float(value.detach().float().reshape(-1)[0].cpu())
    return float(value)


class _ScopedSummary(Summary):
    """Writes events for one candidate scope (reference _ScopedSummaryV2,
    summary.py:375-637: per-scope logdirs)."""

    def __init__(self, logdir: Optional[str], scope: Optional[str] = None,
                 namespace: Optional[str] = None, skip_summary: bool = False):
        self._scope = scope
        self._namespace = namespace
        self._skip = skip_summary or logdir is None
        self._step = 0
        self._lock = threading.Lock()
        self._file = None
        self._tb = None
        # TensorBoard event files alongside the JSONL (the reference's UI
        # is TensorBoard — docs/source/tensorboard.md); per-scope dirs so
        # same-named tags overlay across candidates. ADANET_TB_EVENTS=0
        # turns the tfevents stream off.
        self._tb_enabled = os.environ.get("ADANET_TB_EVENTS", "1") not in (
            "0", "false")
        if not self._skip:
            parts = [logdir]
            if namespace:
                parts.append(namespace)
            if scope:
                parts.append(scope)
            self._dir = os.path.join(*parts)
            os.makedirs(self._dir, exist_ok=True)
            self._path = os.path.join(self._dir, "events.jsonl")

    def _tb_writer(self):
        if self._tb is None and self._tb_enabled and not self._skip:
            from adanet_amd.core.tb_writer import TBEventWriter
            self._tb = TBEventWriter(self._dir)
        return self._tb if self._tb_enabled else None

    @property
    def scope(self):
        return self._scope

    @property
    def logdir(self):
        return None if self._skip else self._dir

    def set_step(self, step: int):
        self._step = int(step)

    def _write(self, kind: str, name: str, payload, family: Optional[str]):
        if self._skip:
            return
        tag = "%s/%s" % (family, name) if family else name
        rec = {"wall_time": time.time(), "step": self._step, "tag": tag,
               "kind": kind, "value": payload}
        with self._lock:
            with open(self._path, "a") as f:
                f.write(json.dumps(rec) + "\n")

    def scalar(self, name, tensor, family=None):
        v = _to_scalar(tensor)
        self._write("scalar", name, v, family)
        tb = None if self._skip else self._tb_writer()
        if tb is not None:
            tag = "%s/%s" % (family, name) if family else name
            with self._lock:
                tb.scalar(tag, v, self._step)

    def histogram(self, name, tensor, family=None):
        if self._skip:
            return
        t = tensor.detach().float().flatten() if isinstance(
            tensor, torch.Tensor) else torch.tensor(tensor, dtype=torch.float32)
        if t.numel() == 0:
            return
        payload = {
            "min": float(t.min()), "max": float(t.max()),
            "mean": float(t.mean()), "std": float(t.std()) if t.numel() > 1 else 0.0,
            "count": int(t.numel()),
        }
        self._write("histogram", name, payload, family)
        tb = self._tb_writer()
        if tb is not None:
            tag = "%s/%s" % (family, name) if family else name
            sample = t[:16384].cpu().tolist()
            with self._lock:
                tb.histogram(tag, sample, self._step)

    def image(self, name, tensor, family=None):
        if self._skip:
            return
        shape = list(tensor.shape) if isinstance(tensor, torch.Tensor) else None
        self._write("image", name, {"shape": shape}, family)

    def audio(self, name, tensor, sample_rate, family=None):
        if self._skip:
            return
        self._write("audio", name, {"sample_rate": sample_rate}, family)

    def text(self, name, value, family=None):
        self._write("text", name, str(value), family)
        tb = None if self._skip else self._tb_writer()
        if tb is not None:
            tag = "%s/%s" % (family, name) if family else name
            with self._lock:
                tb.text(tag, str(value), self._step)


def read_events(path_or_dir: str):
    """Loads a scope's events.jsonl (test/analysis helper; the analog of
    testing_utils.check_eventfile_for_keyword, reference
    adanet/core/testing_utils.py:300)."""
    path = path_or_dir
    if os.path.isdir(path):
        path = os.path.join(path, "events.jsonl")
    out = []
    if not os.path.exists(path):
        return out
    with open(path) as f:
        for line in f:
            line = line.strip()
            if line:
                out.append(json.loads(line))
    return out
