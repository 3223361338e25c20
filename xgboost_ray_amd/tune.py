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

import pickle
from typing import Dict, Optional

from xgboost_ray_amd.session import get_actor_rank, put_queue


def _in_tune_session() -> bool:
    try:
        from ray import tune  # noqa

        return tune.is_session_enabled()
    except Exception:
        return False


class TuneReportCheckpointCallback:
    """Per-iteration metric report + checkpoint for Tune trials.

    Runs inside rank-0 actors; marshals a driver-side callable through the
    queue so the Tune session API is only touched on the driver
    (reference tune.py:26-48).
    """

    def __init__(
        self,
        metrics: Optional[Dict] = None,
        filename: str = "checkpoint",
        frequency: int = 5,
    ):
        self._metrics = metrics
        self._filename = filename
        self._frequency = frequency

    def after_iteration(self, booster, iteration: int, evals_log: Dict) -> bool:
        if get_actor_rank() != 0:
            return False
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
        payload = dict(report_dict)

        def _driver_report(payload=payload):
            try:
                from ray import train as ray_train

                ray_train.report(payload)
            except Exception:
                pass

        put_queue(_driver_report)
        return False


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
