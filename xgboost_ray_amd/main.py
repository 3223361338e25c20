"""Driver orchestration: train() / predict() / RayParams.

Re-implements the reference's L4/L5 layers (reference main.py:448-1896):
actor lifecycle, per-attempt communicator bootstrap, the training event
loop with queue draining and failure detection, checkpoint-based retry
accounting, elastic and non-elastic restart policies, and distributed
prediction with result un-sharding.

Differences from the reference are deliberate MI355X-first choices:
- actors are spawned processes pinned to GPUs by HIP_VISIBLE_DEVICES (one
  per MI355X), not Ray actors in a placement group;
- the Rabit tracker (reference main.py:225-324) is replaced by a
  torch.distributed TCP rendezvous on 127.0.0.1 with a fresh port per
  attempt - RCCL over xGMI does the training-time collectives;
- the driver-held checkpoint / retry-loop semantics are preserved
  (boost_rounds_left -= checkpoint.iteration + 1, reference
  main.py:1606-1713).
"""

import multiprocessing as mp
import pickle
import threading
import time
import warnings
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional, Sequence, Tuple, Union

import numpy as np
import torch

from xgboost_ray_amd.actor import (
    ActorError,
    ActorHandle,
    TrainingError,
    TrainingStoppedError,
)
from xgboost_ray_amd.booster import Booster
from xgboost_ray_amd.env import ENV
from xgboost_ray_amd.matrix import RayDMatrix, RayShardingMode, combine_data
from xgboost_ray_amd.util import Future, find_free_port


class RayXGBoostTrainingError(TrainingError):
    pass


# reference-parity aliases (reference main.py error classes)
RayActorError = ActorError
RayXGBoostTrainingStopped = TrainingStoppedError


class RayXGBoostActorAvailable(RuntimeError):
    """A replacement actor became available during elastic training
    (reference elastic.py:136-142)."""


@dataclass
class RayParams:
    """Distributed training parameters (reference main.py:448-504)."""

    num_actors: int = 0
    cpus_per_actor: int = 0
    gpus_per_actor: int = -1
    resources_per_actor: Optional[Dict] = None
    elastic_training: bool = False
    max_failed_actors: int = 0
    max_actor_restarts: int = 0
    checkpoint_frequency: int = 5
    distributed_callbacks: Optional[List] = None
    verbose: Optional[bool] = None
    placement_options: Optional[Dict] = None

    def get_tune_resources(self):
        from xgboost_ray_amd.tune import _get_tune_resources

        return _get_tune_resources(
            num_actors=self.num_actors,
            cpus_per_actor=self.cpus_per_actor,
            gpus_per_actor=max(0, self.gpus_per_actor),
            resources_per_actor=self.resources_per_actor,
            placement_options=self.placement_options,
        )


def _validate_ray_params(ray_params) -> RayParams:
    if ray_params is None:
        ray_params = RayParams()
    elif isinstance(ray_params, dict):
        ray_params = RayParams(**ray_params)
    elif not isinstance(ray_params, RayParams):
        raise ValueError(
            f"`ray_params` must be a `RayParams` instance or a dict, "
            f"got {type(ray_params)}."
        )
    if ray_params.num_actors <= 0:
        raise ValueError(
            "`num_actors` must be set to a value > 0 in `RayParams`."
        )
    elif ray_params.num_actors < 2:
        warnings.warn(
            "`num_actors` in `RayParams` is smaller than 2 "
            "- training will not be distributed."
        )
    return ray_params


@dataclass
class _Checkpoint:
    """Driver-held in-memory checkpoint (reference main.py:507-510)."""

    iteration: int = -1
    value: Optional[bytes] = None


@dataclass
class _TrainingState:
    """Mutable state threaded through training attempts
    (reference main.py:1038-1058)."""

    actors: List[Optional[ActorHandle]]
    queue: Any
    stop_event: Any
    checkpoint: _Checkpoint
    additional_results: Dict
    failed_actor_ranks: set
    training_started_at: float = 0.0
    placement_group: Any = None
    pending_actors: Dict[int, Tuple[ActorHandle, List[Future]]] = field(
        default_factory=dict
    )
    restart_training_at: Optional[float] = None
    last_resource_check_at: float = 0.0


def _autodetect_resources(ray_params: RayParams, use_tree_method: bool):
    """GPU autodetection: gpus_per_actor = 1 when tree_method is gpu_*
    (reference main.py:835-859)."""
    gpus = ray_params.gpus_per_actor
    if gpus == -1:
        gpus = 1 if use_tree_method and torch.cuda.is_available() else 0
        if use_tree_method and not torch.cuda.is_available():
            gpus = 0
    cpus = ray_params.cpus_per_actor or 1
    return cpus, gpus


def _is_gpu_params(params: Dict) -> bool:
    tm = (params or {}).get("tree_method", "") or ""
    dev = (params or {}).get("device", "") or ""
    return tm.startswith("gpu") or dev.startswith("cuda") or dev.startswith("gpu")


def _create_actor(
    rank: int,
    num_actors: int,
    queue,
    stop_event,
    use_gpu: bool,
    distributed_callbacks,
    cpus_per_actor: int = 0,
) -> ActorHandle:
    handle = ActorHandle(
        rank=rank,
        world_size=num_actors,
        queue=queue,
        stop_event=stop_event,
        config={
            "use_gpu": use_gpu,
            "cpus_per_actor": cpus_per_actor,
            "distributed_callbacks": distributed_callbacks,
        },
        gpu_id=(rank % max(1, torch.cuda.device_count())) if use_gpu else None,
    )
    return handle


def _shard_descriptors(
    dmatrices: List[RayDMatrix], rank: int, num_actors: int
) -> List[Tuple[str, dict]]:
    descs = []
    for dm in dmatrices:
        if dm.distributed:
            descs.append(
                (
                    dm._uid,
                    {
                        "kind": (
                            "streaming"
                            if getattr(dm, "streaming", False)
                            else "distributed"
                        ),
                        "loader": dm.loader,
                        "num_actors": num_actors,
                        "rank": rank,
                    },
                )
            )
        else:
            dm.load_data(num_actors)
            descs.append((dm._uid, {"kind": "refs", "refs": dm.refs[rank]}))
    return descs


def _handle_queue_item(item, state: _TrainingState, callback_returns: Dict):
    """Drain one queue item: checkpoint or callback return
    (reference _handle_queue, main.py:902-922)."""
    rank, payload = item
    if (
        isinstance(payload, tuple)
        and len(payload) == 3
        and payload[0] == "__checkpoint__"
    ):
        _, iteration, value = payload
        state.checkpoint = _Checkpoint(iteration=iteration, value=value)
    elif callable(payload):
        payload()
    else:
        callback_returns.setdefault(rank, []).append(payload)


def _get_actor_alive_status(actors: List[Optional[ActorHandle]]):
    """Probe liveness of all actors (reference elastic.py:145-178)."""
    alive, dead = [], []
    for rank, actor in enumerate(actors):
        if actor is None:
            continue
        if actor.is_alive():
            alive.append(rank)
        else:
            dead.append(rank)
    return alive, dead


def _train(
    params: Dict,
    dtrain: RayDMatrix,
    num_boost_round: int,
    evals: Sequence[Tuple[RayDMatrix, str]],
    ray_params: RayParams,
    use_gpu: bool,
    _training_state: _TrainingState,
    **kwargs,
) -> Tuple[Booster, Dict, Dict]:
    """One training attempt (reference main.py:1061-1337)."""
    from xgboost_ray_amd import elastic

    state = _training_state
    num_actors = ray_params.num_actors

    # -- create missing actors (reference main.py:1129-1149)
    to_create = sorted(state.failed_actor_ranks)
    for rank in to_create:
        if state.actors[rank] is not None:
            raise RuntimeError(
                f"Trying to create actor of rank {rank}, but it already exists."
            )
        actor = _create_actor(
            rank,
            num_actors,
            state.queue,
            state.stop_event,
            use_gpu,
            ray_params.distributed_callbacks,
            cpus_per_actor=ray_params.cpus_per_actor,
        )
        state.actors[rank] = actor
    # start in parallel threads (spawn + torch import takes seconds)
    start_threads = []
    start_errors = []

    def _start(actor):
        try:
            actor.start()
        except Exception as e:  # noqa
            start_errors.append((actor.rank, e))

    for rank in to_create:
        th = threading.Thread(target=_start, args=(state.actors[rank],))
        th.start()
        start_threads.append(th)
    for th in start_threads:
        th.join()
    if start_errors:
        rank, err = start_errors[0]
        state.actors[rank] = None
        raise ActorError(f"Actor {rank} failed to start: {err}")
    state.failed_actor_ranks.clear()

    alive_ranks = [r for r, a in enumerate(state.actors) if a is not None]
    alive_actors = len(alive_ranks)

    # -- data loading (reference main.py:1169-1197)
    dmatrices = [dtrain] + [dm for dm, _ in evals]
    load_futures = []
    for rank in alive_ranks:
        descs = _shard_descriptors(dmatrices, rank, num_actors)
        load_futures.append(state.actors[rank].remote("load_data", descs))
    for fut in load_futures:
        fut.result(timeout=ENV.ACTOR_START_TIMEOUT_S * 4)

    # -- communicator bootstrap (replaces the Rabit tracker,
    #    reference main.py:1207): fresh port, world = alive actors
    master_port = find_free_port()
    comm_rank_of = {r: i for i, r in enumerate(alive_ranks)}

    # -- checkpoint load (reference main.py:1211-1220)
    train_kwargs = dict(kwargs)
    if state.checkpoint.value is not None and state.checkpoint.iteration != -1:
        train_kwargs["xgb_model"] = state.checkpoint.value
    train_kwargs["num_boost_round"] = num_boost_round
    train_kwargs["checkpoint_frequency"] = ray_params.checkpoint_frequency

    # -- launch training (reference main.py:1233-1239)
    state.stop_event.clear()
    evals_spec = [(dm._uid, name) for dm, name in evals]
    train_futures: Dict[int, Future] = {}
    for rank in alive_ranks:
        comm_args = {
            "rank": comm_rank_of[rank],
            "world_size": alive_actors,
            "master_addr": ENV.MASTER_ADDR,
            "master_port": master_port,
        }
        train_futures[rank] = state.actors[rank].remote(
            "train",
            comm_args,
            params,
            dtrain._uid,
            evals_spec,
            rank == alive_ranks[0],
            train_kwargs,
        )

    state.training_started_at = time.time()
    callback_returns: Dict[int, list] = state.additional_results.setdefault(
        "callback_returns", {}
    )
    last_status = time.time()

    # -- driver event loop (reference main.py:1255-1300)
    try:
        while True:
            while not state.queue.empty():
                try:
                    item = state.queue.get_nowait()
                except Exception:
                    break
                _handle_queue_item(item, state, callback_returns)

            if ray_params.elastic_training and not ENV.ELASTIC_RESTART_DISABLED:
                elastic._maybe_schedule_new_actors(
                    training_state=state,
                    ray_params=ray_params,
                    use_gpu=use_gpu,
                    load_data=dmatrices,
                )
                elastic._update_scheduled_actor_states(state)

            pending = [f for f in train_futures.values() if not f.done()]
            if not pending:
                break
            if time.time() - last_status > ENV.STATUS_FREQUENCY_S:
                elapsed = time.time() - state.training_started_at
                if ray_params.verbose:
                    print(
                        f"Training in progress "
                        f"({elapsed:.0f} seconds since last restart)."
                    )
                last_status = time.time()
            # surface errors early
            for rank, fut in train_futures.items():
                if fut.done() and fut._error is not None:
                    raise fut._error
            # liveness probe
            _, dead = _get_actor_alive_status(state.actors)
            if dead:
                raise ActorError(f"Actor(s) {dead} died during training")
            time.sleep(ENV.EVENT_LOOP_POLL_S)

        # collect results; re-raises actor exceptions
        results = {r: f.result() for r, f in train_futures.items()}
        # final drain: queue items (last checkpoint, after_training
        # reports, callback returns) ride the mp.Queue feeder thread, a
        # DIFFERENT channel than the RPC pipe the futures completed on -
        # items put before the actor returned can surface here AFTER the
        # future resolves, and queue.empty() can read stale-True. Drain
        # until the queue stays quiet.
        _drain_queue_settled(state, callback_returns)
    except (ActorError, TrainingError, TrainingStoppedError,
            RayXGBoostActorAvailable) as err:
        # failure path (reference main.py:1302-1316)
        state.stop_event.set()
        # give surviving actors a moment to stop cooperatively
        _wait_all_settled(train_futures, timeout=30.0)
        alive, dead = _get_actor_alive_status(state.actors)
        for rank in dead:
            try:
                state.actors[rank].kill()
            except Exception:
                pass
            state.actors[rank] = None
            state.failed_actor_ranks.add(rank)
        # drain remaining queue items (late checkpoints) - settled drain:
        # the dying rank-0's last checkpoint may still be in the feeder
        _drain_queue_settled(state, callback_returns)
        if isinstance(err, RayXGBoostActorAvailable):
            raise
        raise ActorError(str(err)) from err

    # -- merge results (reference main.py:1322-1337)
    first = results[alive_ranks[0]]
    bst_payload = first["bst"]
    bst = bst_payload
    evals_result = first["evals_result"]
    total_n = sum(res["train_n"] for res in results.values())
    state.additional_results["total_n"] = total_n
    return bst, evals_result, state.additional_results


def _drain_queue_settled(
    state: "_TrainingState",
    callback_returns: Dict,
    settle_s: float = 0.25,
    max_s: float = 3.0,
):
    """Drain the actor queue until it stays empty for ``settle_s``.

    mp.Queue delivery is asynchronous (feeder thread); a plain
    empty()/get_nowait() loop can finish while items are still in
    flight from an actor whose RPC already returned."""
    import queue as _q

    deadline = time.monotonic() + max_s
    last_item = time.monotonic()
    while time.monotonic() < deadline:
        try:
            item = state.queue.get(timeout=0.05)
        except _q.Empty:
            if time.monotonic() - last_item > settle_s:
                return
            continue
        except Exception:
            return
        try:
            _handle_queue_item(item, state, callback_returns)
        except Exception:
            pass
        last_item = time.monotonic()


def _wait_all_settled(train_futures: Dict[int, Future], timeout: float):
    deadline = time.monotonic() + timeout
    for fut in train_futures.values():
        remaining = deadline - time.monotonic()
        if remaining <= 0:
            return
        try:
            fut.result(timeout=remaining)
        except Exception:
            pass


def train(
    params: Dict,
    dtrain: RayDMatrix,
    num_boost_round: int = 10,
    *,
    evals: Union[List[Tuple[RayDMatrix, str]], Tuple] = (),
    evals_result: Optional[Dict] = None,
    additional_results: Optional[Dict] = None,
    ray_params: Union[None, RayParams, Dict] = None,
    _remote: Optional[bool] = None,
    **kwargs,
) -> Booster:
    """Distributed training entry point (reference main.py:1340-1747)."""
    start_time = time.time()
    ray_params = _validate_ray_params(ray_params)

    # fail fast on unknown keyword arguments instead of silently ignoring
    # them inside the actors (reference testKwargsValidation,
    # test_end_to_end.py:355 expects a TypeError naming the kwarg)
    _ALLOWED_TRAIN_KWARGS = {
        "early_stopping_rounds", "verbose_eval", "obj", "feval",
        "maximize", "xgb_model", "callbacks", "num_boost_round",
        "checkpoint_frequency",
    }
    unknown = set(kwargs) - _ALLOWED_TRAIN_KWARGS
    if unknown:
        raise TypeError(
            "train() got unexpected keyword argument(s): "
            + ", ".join(sorted(unknown))
        )

    if not isinstance(dtrain, RayDMatrix):
        raise ValueError(
            f"The `dtrain` argument must be a RayDMatrix, got {type(dtrain)}."
            " Create one with `RayDMatrix(data, label)`."
        )
    for (deval, _name) in evals:
        if not isinstance(deval, RayDMatrix):
            raise ValueError(
                "Evaluation data must be a RayDMatrix, got "
                f"{type(deval)}."
            )

    # Tune integration: auto-inject the report callback when inside a
    # Tune session (reference main.py:1477-1492)
    from xgboost_ray_amd.tune import _try_add_tune_callback

    _try_add_tune_callback(kwargs)

    # unsupported tree methods (reference main.py:1508-1514)
    tm = (params or {}).get("tree_method", "hist") or "hist"
    if tm in ("exact", "approx") or "colmaker" in str(
        (params or {}).get("updater", "")
    ):
        raise ValueError(
            f"`tree_method={tm}` is not supported for distributed training: "
            "column-major / exact greedy updaters cannot be row-sharded. "
            "Use `hist` or `gpu_hist`."
        )

    use_gpu_params = _is_gpu_params(params)
    cpus, gpus = _autodetect_resources(ray_params, use_gpu_params)
    use_gpu = gpus > 0
    if use_gpu_params and not use_gpu:
        warnings.warn(
            "GPU tree method requested but no GPU is available - "
            "training on CPU."
        )

    if ray_params.elastic_training and ray_params.max_failed_actors == 0:
        raise ValueError(
            "Elastic training enabled but `max_failed_actors` is 0. Set "
            "`max_failed_actors` to the number of actors that are allowed "
            "to die without aborting training."
        )
    if ray_params.elastic_training and kwargs.get("early_stopping_rounds"):
        raise ValueError(
            "Early stopping is not supported with elastic training "
            "(inconsistent eval histories across restarts)."
        )

    if isinstance(kwargs.get("xgb_model"), Booster):
        kwargs["xgb_model"] = pickle.dumps(kwargs["xgb_model"])

    # central data can be loaded once up front (reference main.py:1556)
    if not dtrain.distributed:
        dtrain.load_data(ray_params.num_actors)

    ctx = mp.get_context("spawn")
    queue = ctx.Queue()
    stop_event = ctx.Event()

    state = _TrainingState(
        actors=[None] * ray_params.num_actors,
        queue=queue,
        stop_event=stop_event,
        checkpoint=_Checkpoint(),
        additional_results={},
        failed_actor_ranks=set(range(ray_params.num_actors)),
    )

    bst = None
    train_evals_result: Dict = {}
    train_additional_results: Dict = {}
    boost_rounds_left = num_boost_round
    tries = 0
    max_actor_restarts = (
        ray_params.max_actor_restarts
        if ray_params.max_actor_restarts >= 0
        else float("inf")
    )
    total_training_time = 0.0

    while tries <= max_actor_restarts:
        if state.checkpoint.iteration == -1 and state.checkpoint.value:
            # training finished in a previous attempt
            break
        try:
            bst, train_evals_result, train_additional_results = _train(
                params,
                dtrain,
                boost_rounds_left,
                evals,
                ray_params,
                use_gpu,
                _training_state=state,
                **kwargs,
            )
            if state.checkpoint.iteration == -1:
                state.checkpoint = _Checkpoint()
            total_training_time += time.time() - state.training_started_at
            break
        except RayXGBoostActorAvailable:
            # elastic: new actors ready - restart without counting a try
            # (reference main.py:1661-1673)
            total_training_time += time.time() - state.training_started_at
            if state.checkpoint.iteration >= 0:
                boost_rounds_left -= state.checkpoint.iteration + 1
                state.checkpoint.iteration = -2  # consumed marker
            if ray_params.verbose:
                print(
                    "A new actor became available - restarting training "
                    "with a larger world."
                )
            continue
        except (ActorError, TrainingError) as err:
            total_training_time += time.time() - state.training_started_at
            alive_count = sum(a is not None for a in state.actors)
            if state.checkpoint.iteration >= 0:
                boost_rounds_left -= state.checkpoint.iteration + 1
                state.checkpoint.iteration = -2
            if ray_params.elastic_training:
                if alive_count < ray_params.num_actors - ray_params.max_failed_actors:
                    raise RuntimeError(
                        f"Too many actors died ({ray_params.num_actors - alive_count}"
                        f" > max_failed_actors={ray_params.max_failed_actors}):"
                        f" aborting. Root cause: {err}"
                    ) from err
                # continue with fewer actors; dead ranks are NOT restarted
                state.failed_actor_ranks.clear()
                if ray_params.verbose:
                    print(
                        f"Continuing elastic training with "
                        f"{alive_count} remaining actors."
                    )
                # elastic failures don't consume a retry
            else:
                if tries >= max_actor_restarts:
                    raise RuntimeError(
                        f"Training failed and max_actor_restarts="
                        f"{ray_params.max_actor_restarts} exhausted."
                    ) from err
                tries += 1
                if ray_params.verbose:
                    print(
                        f"Restarting training from latest checkpoint "
                        f"(attempt {tries}/{ray_params.max_actor_restarts})."
                    )
            continue

    if bst is None and state.checkpoint.value is not None:
        bst = pickle.loads(state.checkpoint.value)

    # shutdown actors (reference main.py:1715-1745)
    for actor in state.actors:
        if actor is not None:
            actor.shutdown()
    for rank, (actor, _futs) in list(state.pending_actors.items()):
        actor.shutdown()
    try:
        queue.close()
    except Exception:
        pass

    if evals_result is not None:
        evals_result.update(train_evals_result)
    train_additional_results["training_time_s"] = total_training_time
    train_additional_results["total_time_s"] = time.time() - start_time
    if additional_results is not None:
        additional_results.update(train_additional_results)
    return bst


def _predict(
    model: Booster,
    data: RayDMatrix,
    ray_params: RayParams,
    use_gpu: bool,
    **kwargs,
) -> np.ndarray:
    """One prediction attempt (reference main.py:1750-1806)."""
    num_actors = ray_params.num_actors
    ctx = mp.get_context("spawn")
    queue = ctx.Queue()
    stop_event = ctx.Event()
    actors = [
        _create_actor(
            rank, num_actors, queue, stop_event, use_gpu,
            ray_params.distributed_callbacks,
        )
        for rank in range(num_actors)
    ]
    try:
        threads = [threading.Thread(target=a.start) for a in actors]
        for t in threads:
            t.start()
        for t in threads:
            t.join()
        for a in actors:
            if not a.is_alive():
                raise ActorError(f"Prediction actor {a.rank} failed to start")
        load_futures = []
        for rank, actor in enumerate(actors):
            descs = _shard_descriptors([data], rank, num_actors)
            load_futures.append(actor.remote("load_data", descs))
        for fut in load_futures:
            fut.result(timeout=ENV.ACTOR_START_TIMEOUT_S * 4)
        # ship the model once (reference main.py:1790 ray.put(model))
        model_bytes = pickle.dumps(model)
        pred_futures = [
            actor.remote("predict", model_bytes, data._uid, kwargs)
            for actor in actors
        ]
        results = [f.result() for f in pred_futures]
    finally:
        for a in actors:
            a.shutdown()
    if data.sharding == RayShardingMode.FIXED:
        return np.concatenate(results, axis=0)
    actor_idx = getattr(getattr(data, "loader", None), "actor_indices", None)
    if actor_idx and data.sharding == RayShardingMode.INTERLEAVED:
        # qid matrices shard whole query groups (not raw interleaved
        # rows) - reassemble through the recorded per-actor indices
        first = np.asarray(results[0])
        n = sum(len(r) for r in results)
        shape = (n,) + first.shape[1:]
        out = np.empty(shape, dtype=first.dtype)
        for rank, part in enumerate(results):
            out[actor_idx[rank]] = np.asarray(part)
        return out
    return combine_data(data.sharding, results)


def predict(
    model: Booster,
    data: RayDMatrix,
    ray_params: Union[None, RayParams, Dict] = None,
    _remote: Optional[bool] = None,
    **kwargs,
) -> Optional[np.ndarray]:
    """Distributed prediction (reference main.py:1809-1896)."""
    ray_params = _validate_ray_params(ray_params)
    _ALLOWED_PREDICT_KWARGS = {
        "output_margin", "iteration_range", "pred_leaf", "pred_contribs",
        "pred_interactions", "approx_contribs", "validate_features",
    }
    unknown = set(kwargs) - _ALLOWED_PREDICT_KWARGS
    if unknown:
        raise TypeError(
            "predict() got unexpected keyword argument(s): "
            + ", ".join(sorted(unknown))
        )
    if not isinstance(data, RayDMatrix):
        raise ValueError(
            f"The `data` argument must be a RayDMatrix, got {type(data)}."
        )
    # GPU routing (MI355X-first): predict on the GPU whenever one is
    # present unless the caller explicitly pins gpus_per_actor=0. The
    # saved model schema does not carry tree_method (stock XGBoost's
    # doesn't either), so inferring from model.params alone sent loaded
    # models through the CPU walker (round-2 finding: 174 s for 5M rows
    # vs 2.3 s routed).
    use_gpu = (
        torch.cuda.is_available() and ray_params.gpus_per_actor != 0
    )
    max_actor_restarts = (
        ray_params.max_actor_restarts
        if ray_params.max_actor_restarts >= 0
        else float("inf")
    )
    tries = 0
    while tries <= max_actor_restarts:
        try:
            return _predict(model, data, ray_params, use_gpu, **kwargs)
        except (ActorError, TrainingError):
            if tries >= max_actor_restarts:
                raise
            tries += 1
    return None
