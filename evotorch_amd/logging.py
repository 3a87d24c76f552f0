"""Logger sinks subscribing to a searcher's log_hook.

Reference parity: /root/reference/src/evotorch/logging.py:67-763
(Logger :67, PicklingLogger :110, StdOutLogger :428, PandasLogger :479,
and the import-guarded Sacred/Mlflow/Neptune/Wandb sinks :525-763).
"""

import os
import pickle
from datetime import datetime
from typing import Optional

import torch

__all__ = ["Logger", "StdOutLogger", "PandasLogger", "PicklingLogger", "MlflowLogger", "SacredLogger", "NeptuneLogger", "WandbLogger"]


class Logger:
    """Base: attaches itself to `searcher.log_hook`; receives the status
    dict each generation."""

    def __init__(self, searcher, *, interval: int = 1, after_first_step: bool = False):
        searcher.log_hook.append(self)
        self._interval = int(interval)
        self._after_first_step = bool(after_first_step)
        self._steps_count = 0

    def __call__(self, status: dict):
        if self._after_first_step:
            n = self._steps_count
            self._steps_count += 1
        else:
            self._steps_count += 1
            n = self._steps_count
        if n % self._interval == 0:
            self._log(self._filter(status))

    def _filter(self, status: dict) -> dict:
        return status

    def _log(self, status: dict):
        raise NotImplementedError


class ScalarLogger(Logger):
    """Keeps only scalar-valued status items (what most sinks can accept)."""

    def _filter(self, status: dict) -> dict:
        out = {}
        for k, v in status.items():
            if isinstance(v, (int, float, bool, str)):
                out[k] = v
            elif isinstance(v, torch.Tensor) and v.ndim == 0:
                out[k] = v.item()
        return out


class StdOutLogger(ScalarLogger):
    """Prints the status table to stdout each logged generation
    (reference logging.py:428)."""

    def __init__(self, searcher, *, interval: int = 1, after_first_step: bool = False, leading_keys: tuple = ("iter",)):
        super().__init__(searcher, interval=interval, after_first_step=after_first_step)
        self._leading_keys = tuple(leading_keys)

    def _log(self, status: dict):
        keys = [k for k in self._leading_keys if k in status] + [k for k in status if k not in self._leading_keys]
        for k in keys:
            print(f"{str(k):>22} : {status[k]}")
        print()


class PandasLogger(ScalarLogger):
    """Accumulates the scalar status into a pandas DataFrame
    (reference logging.py:479)."""

    def __init__(self, searcher, *, interval: int = 1, after_first_step: bool = False):
        super().__init__(searcher, interval=interval, after_first_step=after_first_step)
        self._records = []

    def _log(self, status: dict):
        self._records.append(dict(status))

    def to_dataframe(self, *, index: Optional[str] = "iter"):
        import pandas as pd

        frame = pd.DataFrame(self._records)
        if index is not None and index in frame.columns:
            frame = frame.set_index(index)
        return frame


class PicklingLogger(Logger):
    """Checkpointing: every `interval` generations pickles the requested
    status items (center/best/pop_best by default) plus, for
    neuroevolution problems, a ready-to-run policy module
    (reference logging.py:110-418). Also registers a final save on the
    searcher's end_of_run_hook."""

    def __init__(
        self,
        searcher,
        *,
        interval: int,
        directory: Optional[str] = None,
        prefix: Optional[str] = None,
        zfill: int = 6,
        items_to_save: tuple = ("center", "best", "pop_best", "median_eval", "mean_eval"),
        make_policy_from: Optional[str] = None,
        after_first_step: bool = False,
        verbose: bool = True,
    ):
        super().__init__(searcher, interval=interval, after_first_step=after_first_step)
        self._searcher = searcher
        self._directory = directory or "."
        if directory:
            os.makedirs(directory, exist_ok=True)
        self._prefix = prefix or f"{type(searcher).__name__}_{datetime.now().strftime('%Y%m%d_%H%M%S')}"
        self._zfill = int(zfill)
        self._items = tuple(items_to_save)
        self._make_policy_from = make_policy_from
        self._verbose = bool(verbose)
        self._last_file: Optional[str] = None
        searcher.end_of_run_hook.append(self._final_save)

    @property
    def last_file_name(self) -> Optional[str]:
        return self._last_file

    def _final_save(self, status: dict):
        self.save(status)

    def _log(self, status: dict):
        self.save(status)

    def _policy_item(self, status: dict):
        problem = self._searcher.problem
        if self._make_policy_from is not None:
            source = status.get(self._make_policy_from, None)
        else:
            source = status.get("center", status.get("pop_best", None))
        if source is None or not hasattr(problem, "to_policy"):
            return None
        try:
            from .core import Solution

            x = source.values if isinstance(source, Solution) else source
            return problem.to_policy(torch.Tensor.as_subclass(torch.as_tensor(x), torch.Tensor))
        except Exception:
            return None

    def save(self, status: Optional[dict] = None) -> str:
        if status is None:
            status = dict(self._searcher.status)
        payload = {}
        for k in self._items:
            if k in status:
                v = status[k]
                if isinstance(v, torch.Tensor):
                    v = torch.Tensor.as_subclass(v, torch.Tensor).detach().cpu().clone()
                payload[k] = v
        policy = self._policy_item(status)
        if policy is not None:
            payload["policy"] = policy
        problem = self._searcher.problem
        if hasattr(problem, "observation_normalization_data"):
            try:
                payload["obs_norm"] = problem.observation_normalization_data()
            except Exception:
                pass
        fname = os.path.join(self._directory, f"{self._prefix}_generation{str(self._steps_count).zfill(self._zfill)}.pickle")
        with open(fname, "wb") as f:
            pickle.dump(payload, f)
        self._last_file = fname
        if self._verbose:
            print(f"[PicklingLogger] saved {fname}")
        return fname

    def unpickle_last_file(self):
        with open(self._last_file, "rb") as f:
            return pickle.load(f)


def _require(modname: str):
    import importlib

    try:
        return importlib.import_module(modname)
    except ImportError as e:
        raise ImportError(f"{modname} is required for this logger but is not installed") from e


class MlflowLogger(ScalarLogger):
    """Logs scalar metrics to an MLflow run (reference logging.py:573)."""

    def __init__(self, searcher, client=None, run=None, *, interval: int = 1, after_first_step: bool = False):
        super().__init__(searcher, interval=interval, after_first_step=after_first_step)
        mlflow = _require("mlflow")
        self._client = client if client is not None else mlflow.tracking.MlflowClient()
        self._run_id = run.info.run_id if run is not None else mlflow.active_run().info.run_id

    def _log(self, status: dict):
        for k, v in status.items():
            if isinstance(v, (int, float)):
                self._client.log_metric(self._run_id, k, v)


class SacredLogger(ScalarLogger):
    """Logs scalar metrics to a sacred Run (reference logging.py:525)."""

    def __init__(self, searcher, run, result: Optional[str] = None, *, interval: int = 1, after_first_step: bool = False):
        super().__init__(searcher, interval=interval, after_first_step=after_first_step)
        self._run = run
        self._result = result

    def _log(self, status: dict):
        for k, v in status.items():
            if isinstance(v, (int, float)):
                self._run.log_scalar(k, v)
        if self._result is not None and self._result in status:
            self._run.result = status[self._result]


class NeptuneLogger(ScalarLogger):
    """Logs scalar metrics to a neptune run (reference logging.py:636)."""

    def __init__(self, searcher, run=None, *, interval: int = 1, after_first_step: bool = False, group: Optional[str] = None, **neptune_kwargs):
        super().__init__(searcher, interval=interval, after_first_step=after_first_step)
        if run is None:
            neptune = _require("neptune")
            run = neptune.init_run(**neptune_kwargs)
        self._run = run
        self._group = group

    @property
    def run(self):
        return self._run

    def _log(self, status: dict):
        for k, v in status.items():
            key = k if self._group is None else f"{self._group}/{k}"
            self._run[key].log(v)


class WandbLogger(ScalarLogger):
    """Logs scalar metrics to Weights & Biases (reference logging.py:696)."""

    def __init__(self, searcher, init: bool = True, *, interval: int = 1, after_first_step: bool = False, **wandb_kwargs):
        super().__init__(searcher, interval=interval, after_first_step=after_first_step)
        self._wandb = _require("wandb")
        if init:
            self._wandb.init(**wandb_kwargs)

    def _log(self, status: dict):
        self._wandb.log(status)
