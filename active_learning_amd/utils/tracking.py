"""Experiment tracking with the reference's Comet metric-name contract.

The reference logs everything to Comet ML (src/main_al.py:101-114, docstring
src/main_al.py:24-40 lists the metric names; src/query_strategies/strategy.py
logs rd_test_accuracy / budget_test_accuracy / rd_{r}_validation_accuracy /
cumulative_budget and assets labeled_idxs_on_rd_{r} / test_acc_byclass_rd_{r}).

This module provides a local JSONL-backed tracker with a Comet-compatible API
surface; when comet_ml is importable and --enable_comet is passed, metrics are
mirrored to a real Comet experiment as well. The experiment key/hash contract
(hash keys the checkpoint directory, src/main_al.py:107-111) is preserved.
"""

import json
import os
import time
import uuid

try:  # optional — not installed in this environment
    import comet_ml  # type: ignore
    _HAS_COMET = True
except Exception:  # pragma: no cover
    comet_ml = None
    _HAS_COMET = False


class Experiment:
    """Local experiment tracker; drop-in for the subset of comet_ml.Experiment
    the reference uses (log_metric(s), log_parameters, log_asset_data,
    add_tag, set_name, get_key, .url)."""

    def __init__(self, project_name="active-learning", disabled=False, log_dir="./logs",
                 experiment_key=None, mirror_comet=True, **_ignored):
        self.project_name = project_name
        self.disabled = disabled
        # spawned trainer ranks re-attach ExistingExperiment without the CLI
        # args in scope (strategy.py:290-293); the parent exports the real
        # log dir so the rank-0 JSONL lands in the same place. Only the
        # DEFAULT is overridden — explicit log_dir arguments win.
        if log_dir == "./logs":
            log_dir = os.environ.get("AL_TRACK_LOG_DIR", log_dir)
        self.log_dir = log_dir
        self.key = experiment_key or uuid.uuid4().hex[:16]
        self.name = None
        self.tags = []
        self._fh = None
        self._comet = None
        if not disabled and mirror_comet and _HAS_COMET:
            try:  # pragma: no cover — comet not present in CI
                self._comet = comet_ml.Experiment(project_name=project_name,
                                                  auto_param_logging=False,
                                                  auto_metric_logging=False)
                self.key = self._comet.get_key()
            except Exception:
                self._comet = None

    # -- identity -----------------------------------------------------------
    @property
    def url(self):
        # The reference derives exp_hash from basename(url)[:9] (main_al.py:107).
        if self._comet is not None:  # pragma: no cover
            return self._comet.url
        return f"local://experiments/{self.key}"

    def get_key(self):
        return self.key

    def set_name(self, name):
        self.name = name
        if self._comet:  # pragma: no cover
            self._comet.set_name(name)

    def add_tag(self, tag):
        self.tags.append(tag)
        if self._comet:  # pragma: no cover
            self._comet.add_tag(tag)

    # -- sinks --------------------------------------------------------------
    def _write(self, record):
        if self.disabled:
            return
        if self._fh is None:
            os.makedirs(self.log_dir, exist_ok=True)
            path = os.path.join(self.log_dir, f"metrics_{self.key}.jsonl")
            self._fh = open(path, "a")
        record["t"] = time.time()
        self._fh.write(json.dumps(record, default=str) + "\n")
        self._fh.flush()

    def log_parameters(self, params):
        self._write({"kind": "parameters", "params": dict(params)})
        if self._comet:  # pragma: no cover
            self._comet.log_parameters(params)

    def log_metric(self, name, value, step=None, include_context=True, **_):
        self._write({"kind": "metric", "name": name, "value": _to_py(value), "step": _to_py(step)})
        if self._comet:  # pragma: no cover
            self._comet.log_metric(name, value, step=step)

    def log_metrics(self, metrics, step=None, **_):
        for k, v in metrics.items():
            self.log_metric(k, v, step=step)

    def log_asset_data(self, data, name=None, **_):
        self._write({"kind": "asset", "name": name, "data": data if isinstance(data, str)
                     else str(data)})
        if self._comet:  # pragma: no cover
            self._comet.log_asset_data(data, name=name)

    # -- pickling: drop file handles/comet (strategy pickles exclude the
    # experiment anyway, matching resume_training.py:45-52) ------------------
    def __getstate__(self):
        state = dict(self.__dict__)
        state["_fh"] = None
        state["_comet"] = None
        return state


class ExistingExperiment(Experiment):
    """Re-attach to a previous experiment by key (resume path).

    Reference: comet_ml.ExistingExperiment at src/utils/resume_training.py:29-32.
    """

    def __init__(self, previous_experiment=None, log_dir="./logs", **kwargs):
        super().__init__(experiment_key=previous_experiment, log_dir=log_dir, **kwargs)


def _to_py(v):
    try:
        import torch
        if isinstance(v, torch.Tensor):
            return v.item() if v.numel() == 1 else v.tolist()
    except Exception:
        pass
    try:
        import numpy as np
        if isinstance(v, np.generic):
            return v.item()
    except Exception:
        pass
    return v
