"""Runtime flag system, overridable by ``RXGB_<NAME>`` environment variables.

Mirrors the reference's ``_XGBoostEnv`` singleton (reference main.py:110-162):
every attribute read is re-checked against the environment with type coercion
from the default value's type.
"""

import os
from dataclasses import dataclass


def _get_environ(item: str, old_val):
    env_var = f"RXGB_{item.upper()}"
    new_val = os.environ.get(env_var)
    if new_val is None:
        return old_val
    if isinstance(old_val, bool):
        return new_val.lower() in ("1", "true", "yes")
    return type(old_val)(new_val)


@dataclass
class _Env:
    # How often the driver logs "Training in progress" status lines (seconds).
    STATUS_FREQUENCY_S: int = 30
    # Elastic training: disable in-train actor restarts entirely.
    ELASTIC_RESTART_DISABLED: bool = False
    # Elastic training: how often to check whether new actors can be scheduled.
    ELASTIC_RESTART_RESOURCE_CHECK_S: int = 30
    # Elastic training: grace period before restarting with recovered actors.
    ELASTIC_RESTART_GRACE_PERIOD_S: int = 10
    # Seconds to wait for actor processes to start before raising.
    ACTOR_START_TIMEOUT_S: int = 120
    # Seconds to wait when gracefully shutting an actor down before killing it.
    ACTOR_SHUTDOWN_TIMEOUT_S: int = 10
    # torch.distributed rendezvous address for the per-attempt communicator.
    MASTER_ADDR: str = "127.0.0.1"
    # Driver polling interval in the training event loop (seconds).
    EVENT_LOOP_POLL_S: float = 0.1

    def __getattribute__(self, item):
        old_val = object.__getattribute__(self, item)
        if item.startswith("_") or callable(old_val):
            return old_val
        return _get_environ(item, old_val)


ENV = _Env()
