"""Ray Tune integration hooks (reference tune.py:1-156).

Ray is not a dependency of this framework; the integration surface is kept
so code written against the reference keeps working when Ray IS installed:
- ``_try_add_tune_callback`` auto-injects a report callback inside a Tune
  session (no-op otherwise, reference tune.py:60-104);
- ``TuneReportCheckpointCallback`` reports per-round metrics + checkpoints
  through the driver queue (callables executed driver-side, reference
  tune.py:26-48);
- ``_get_tune_resources`` builds the placement-group factory equivalent.
"""

import os
import pickle
from typing import Dict, Optional

from xgboost_ray_amd.session import get_actor_rank, put_queue


def _in_tune_session() -> bool:
    try:
        from ray import tune  # noqa

        return tune.is_session_enabled()
    except Exception:
        return False


class _DriverReportAndCheckpoint:
    """Driver-side executable queue payload.

    Must be a module-level class: the queue is a multiprocessing.Queue
    whose feeder thread pickles items, and closures/lambdas are not
    picklable (they would be dropped silently in the feeder thread).

    On the driver it (1) writes the serialized Booster into
    ``<trial_dir>/checkpoint_<iteration>/<filename>`` when a checkpoint
    is due, and (2) reports the metrics (with the checkpoint attached)
    through ``ray.train.report`` when Ray is importable — mirroring the
    upstream ray.tune.integration.xgboost callback the reference wraps
    (reference tune.py:26-48). Without Ray, the trial dir comes from the
    ``results_dir`` callback argument or ``RXGB_TUNE_RESULT_DIR``.
    """

    def __init__(self, metrics, raw_model, iteration, filename, results_dir):
        self.metrics = metrics
        self.raw_model = raw_model
        self.iteration = iteration
        self.filename = filename
        self.results_dir = results_dir

    def _write_local(self, base: str) -> str:
        ckpt_dir = os.path.join(
            base, f"checkpoint_{self.iteration:06d}"
        )
        os.makedirs(ckpt_dir, exist_ok=True)
        path = os.path.join(ckpt_dir, self.filename)
        with open(path, "wb") as f:
            f.write(self.raw_model)
        return ckpt_dir

    def __call__(self):
        checkpoint = None
        if self.raw_model is not None:
            base = self.results_dir or os.environ.get(
                "RXGB_TUNE_RESULT_DIR"
            )
            ckpt_dir = None
            if base is None:
                try:
                    from ray import train as ray_train

                    base = ray_train.get_context().get_trial_dir()
                except Exception:
                    base = None
            if base is not None:
                ckpt_dir = self._write_local(base)
            if ckpt_dir is not None:
                try:
                    from ray.train import Checkpoint

                    checkpoint = Checkpoint.from_directory(ckpt_dir)
                except Exception:
                    checkpoint = None
        try:
            from ray import train as ray_train

            ray_train.report(dict(self.metrics), checkpoint=checkpoint)
        except Exception:
            pass


class TuneReportCheckpointCallback:
    """Per-iteration metric report + checkpoint for Tune trials.

    Runs inside rank-0 actors; marshals a driver-side payload through the
    queue so the Tune session API is only touched on the driver
    (reference tune.py:26-48). ``filename``/``frequency`` are honored: a
    checkpoint file is written every ``frequency`` completed iterations
    (and once more after training); ``frequency=0`` reports metrics only.
    ``results_dir`` overrides the trial directory when running without
    Ray (tests, plain multiprocess runs).
    """

    def __init__(
        self,
        metrics: Optional[Dict] = None,
        filename: str = "checkpoint",
        frequency: int = 5,
        results_dir: Optional[str] = None,
    ):
        self._metrics = metrics
        self._filename = filename
        self._frequency = frequency
        self._results_dir = results_dir
        self._last_ckpt_iter = -1

    def _report_dict(self, evals_log: Dict) -> Dict:
        report_dict = {}
        for ename, metrics in (evals_log or {}).items():
            for mname, values in metrics.items():
                key = f"{ename}-{mname}"
                report_dict[key] = values[-1]
        if self._metrics:
            if isinstance(self._metrics, dict):
                report_dict = {
                    k: report_dict.get(v, None) for k, v in self._metrics.items()
                }
            else:
                report_dict = {
                    k: report_dict.get(k) for k in self._metrics
                }
        return dict(report_dict)

    def _put(self, booster, iteration: int, evals_log: Dict, force: bool):
        due = force or (
            self._frequency > 0
            and (iteration + 1) % self._frequency == 0
        )
        raw = None
        if due and iteration != self._last_ckpt_iter:
            raw = booster.save_raw("ubj")
            self._last_ckpt_iter = iteration
        put_queue(
            _DriverReportAndCheckpoint(
                self._report_dict(evals_log),
                raw,
                iteration,
                self._filename,
                self._results_dir,
            )
        )

    def after_iteration(self, booster, iteration: int, evals_log: Dict) -> bool:
        if get_actor_rank() != 0:
            return False
        self._put(booster, iteration, evals_log, force=False)
        return False

    def after_training(self, booster, iteration: int, evals_log: Dict):
        """Final checkpoint regardless of frequency (upstream semantics)."""
        if get_actor_rank() != 0:
            return
        self._put(booster, iteration, evals_log, force=True)


def _try_add_tune_callback(kwargs: Dict) -> bool:
    """Inside a Tune session, ensure a report callback is present
    (reference tune.py:60-104)."""
    if not _in_tune_session():
        return False
    callbacks = list(kwargs.get("callbacks") or [])
    if not any(isinstance(cb, TuneReportCheckpointCallback) for cb in callbacks):
        callbacks.append(TuneReportCheckpointCallback())
    kwargs["callbacks"] = callbacks
    return True


def _get_tune_resources(
    num_actors: int,
    cpus_per_actor: int,
    gpus_per_actor: int,
    resources_per_actor: Optional[Dict] = None,
    placement_options: Optional[Dict] = None,
):
    """PlacementGroupFactory equivalent (reference tune.py:107-126)."""
    try:
        from ray.tune import PlacementGroupFactory
    except Exception as e:
        raise RuntimeError(
            "Tune resources require `ray` to be installed."
        ) from e
    head = {"CPU": 1}
    child = {"CPU": cpus_per_actor, "GPU": gpus_per_actor}
    if resources_per_actor:
        child.update(resources_per_actor)
    bundles = [head] + [child] * num_actors
    options = dict(placement_options or {})
    options.setdefault("strategy", "PACK")
    return PlacementGroupFactory(bundles, **options)


def load_model(model_path: str):
    """Load a Booster from a Tune checkpoint path (reference tune.py:130-156)."""
    from xgboost_ray_amd.booster import Booster

    bst = Booster()
    try:
        bst.load_model(model_path)
    except Exception:
        with open(model_path, "rb") as f:
            bst = pickle.load(f)
    return bst
