"""Training actors: one process per MI355X GPU.

MI355X-native replacement for ``RayXGBoostActor`` + Ray's C++ actor
runtime (reference main.py:543-815, SURVEY.md #2.3 'Ray core'): each actor
is a spawned process pinned to one GPU via ``HIP_VISIBLE_DEVICES``, driven
over a duplex pipe with Future-based async calls (the ``.remote()``
semantics the reference gets from Ray), with a shared queue for
checkpoints/callback returns and a shared stop event polled every boosting
round (the reference polls it in a daemon thread, main.py:774-781; our
engine checks it in an after-iteration callback, so no thread is needed).
"""

import os
import pickle
import threading
import time
import traceback
import multiprocessing as mp
from typing import Any, Dict, List, Optional, Tuple

import numpy as np

from xgboost_ray_amd import session
from xgboost_ray_amd.callback import DistributedCallbackContainer
from xgboost_ray_amd.env import ENV


class ActorError(RuntimeError):
    """An actor process died (RayActorError equivalent)."""


class TrainingStoppedError(RuntimeError):
    """Training was interrupted via the stop event
    (reference RayXGBoostTrainingStopped)."""


class TrainingError(RuntimeError):
    """Wraps an exception raised inside an actor
    (reference RayXGBoostTrainingError)."""


# --------------------------------------------------------------------------
# Actor-process side
# --------------------------------------------------------------------------


class _ActorWorker:
    """Executes commands inside the actor process."""

    def __init__(self, rank: int, world_size: int, queue, stop_event, config):
        self.rank = rank
        self.world_size = world_size
        self.queue = queue
        self.stop_event = stop_event
        self.config = config or {}
        self.device = self._setup_device()
        self.shards: Dict[str, Dict[str, Any]] = {}  # dmatrix uid -> shard
        self.dist_callbacks = DistributedCallbackContainer(
            self.config.get("distributed_callbacks")
        )
        session.init_session(rank, world_size, queue)
        self.dist_callbacks.on_init(self)
        self._stop_generation = 0

    def _setup_device(self):
        import torch

        # intra-actor thread parallelism (reference _set_omp_num_threads,
        # main.py:355-362): torch CPU threads = cpus_per_actor
        cpus = int(self.config.get("cpus_per_actor", 0) or 0)
        if cpus > 0:
            torch.set_num_threads(cpus)
            os.environ["OMP_NUM_THREADS"] = str(cpus)
        use_gpu = self.config.get("use_gpu", False)
        if use_gpu:
            if not torch.cuda.is_available():
                raise RuntimeError(
                    "Actor configured for GPU but torch.cuda is unavailable "
                    "(HIP runtime / MI355X not found)"
                )
            torch.cuda.set_device(0)  # HIP_VISIBLE_DEVICES pins the GPU
            return torch.device("cuda", 0)
        return torch.device("cpu")

    # -- RPC methods -------------------------------------------------------
    def pid(self):
        return os.getpid()

    def ip(self):
        return "127.0.0.1"

    def ping(self):
        return "pong"

    def set_queue(self, queue=None):
        # queue is shared at spawn; this (re)attaches it to the session
        session.set_session_queue(self.queue)
        return True

    def set_stop_generation(self, gen: int):
        """Guards against a stale stop event after restart
        (reference main.py:630-641 stale-event check)."""
        self._stop_generation = gen
        return True

    def load_data(self, descriptors: List[Tuple[str, dict]]):
        """Load this actor's shard of each matrix (uid -> descriptor)."""
        import torch

        for uid, desc in descriptors:
            if uid in self.shards:
                continue
            self.dist_callbacks.before_data_loading(self, desc)
            shard = _materialize_shard(desc)
            self.shards[uid] = shard
            self.dist_callbacks.after_data_loading(self, desc)
        return {
            uid: (int(s["data"].shape[0]) if s.get("data") is not None else -1)
            for uid, s in self.shards.items()
        }

    def train(
        self,
        comm_args: Dict[str, Any],
        params: Dict,
        dtrain_uid: str,
        evals: List[Tuple[str, str]],
        return_bst: bool,
        kwargs: Dict[str, Any],
    ) -> Dict[str, Any]:
        import torch

        from xgboost_ray_amd.engine.collective import Collective
        from xgboost_ray_amd.engine.quantile import BinnedMatrix
        from xgboost_ray_amd.engine.trainer import EvalPack, run_training

        self.dist_callbacks.before_train(self)
        shard = self.shards.get(dtrain_uid)
        if shard is None:
            raise RuntimeError(f"Training data {dtrain_uid} not loaded on actor")

        coll = Collective(
            rank=comm_args["rank"],
            world_size=comm_args["world_size"],
            master_addr=comm_args.get("master_addr", ENV.MASTER_ADDR),
            master_port=comm_args["master_port"],
            device=self.device,
        )
        try:
            if shard.get("streaming"):
                if params.get("booster") == "gblinear":
                    raise ValueError(
                        "booster=gblinear is not supported with streaming "
                        "(RayDeviceQuantileDMatrix) training data"
                    )
                loader = shard["loader"]

                def chunk_fn(loader=loader, shard=shard):
                    return loader.iter_shards(
                        shard["rank"], shard["num_actors"]
                    )

                dm = BinnedMatrix.build_streaming(
                    chunk_fn,
                    n_features=0,
                    device=self.device,
                    max_bin=int(params.get("max_bin", 256)),
                    collective=coll,
                )
            else:
                t_x = torch.from_numpy(
                    np.ascontiguousarray(shard["data"], dtype=np.float32)
                ).to(self.device)
                dm = BinnedMatrix.build(
                    t_x,
                    label=_to_dev(shard.get("label"), self.device),
                    weight=_to_dev(shard.get("weight"), self.device),
                    base_margin=_to_dev(shard.get("base_margin"), self.device),
                    qid=_to_dev(shard.get("qid"), self.device, dtype=None),
                    max_bin=int(params.get("max_bin", 256)),
                    collective=coll,
                    seed=int(params.get("seed", 0) or 0),
                    keep_raw=params.get("booster") == "gblinear",
                )
                del t_x
                if shard.get("feature_weights") is not None:
                    dm.feature_weights = _to_dev(
                        shard["feature_weights"], self.device
                    )
                for bkey in ("label_lower_bound", "label_upper_bound"):
                    if shard.get(bkey) is not None:
                        setattr(
                            dm, bkey, _to_dev(shard[bkey], self.device)
                        )

            eval_packs = []
            for uid, name in evals:
                if uid == dtrain_uid:
                    eval_packs.append(EvalPack(name=name, X=None))
                else:
                    es = self.shards[uid]
                    if es.get("streaming"):
                        raise ValueError(
                            "Streaming (RayDeviceQuantileDMatrix) eval sets "
                            "are not supported - evaluate on the training "
                            "matrix or an in-memory RayDMatrix."
                        )
                    eval_packs.append(
                        EvalPack(
                            name=name,
                            X=torch.from_numpy(
                                np.ascontiguousarray(es["data"], np.float32)
                            ).to(self.device),
                            label=_to_dev(es.get("label"), self.device),
                            weight=_to_dev(es.get("weight"), self.device),
                            qid=_to_dev(es.get("qid"), self.device, dtype=None),
                            base_margin=_to_dev(
                                es.get("base_margin"), self.device
                            ),
                        )
                    )

            xgb_model = kwargs.get("xgb_model")
            if isinstance(xgb_model, (bytes, bytearray)):
                xgb_model = pickle.loads(xgb_model)

            stop_cb = _StopCallback(self.stop_event)
            callbacks = list(kwargs.get("callbacks") or [])
            callbacks.append(stop_cb)
            if self.rank == 0:
                callbacks.append(
                    _CheckpointCallback(
                        frequency=int(kwargs.get("checkpoint_frequency", 5) or 0),
                        queue=self.queue,
                        rank=self.rank,
                    )
                )

            evals_result: Dict = {}
            bst = run_training(
                params,
                dm,
                int(kwargs.get("num_boost_round", 10)),
                evals=eval_packs,
                collective=coll,
                rank=comm_args["rank"],
                xgb_model=xgb_model,
                callbacks=callbacks,
                early_stopping_rounds=kwargs.get("early_stopping_rounds"),
                verbose_eval=kwargs.get("verbose_eval", False),
                obj=kwargs.get("obj"),
                feval=kwargs.get("feval"),
                maximize=kwargs.get("maximize"),
                evals_result=evals_result,
            )
            if stop_cb.stopped:
                raise TrainingStoppedError("Training was interrupted")
            if self.rank == 0:
                # final checkpoint: iteration == -1 marks finished training
                # (reference main.py:621-624)
                self.queue.put(
                    (self.rank, ("__checkpoint__", -1, pickle.dumps(bst)))
                )
            result = {
                "bst": bst if return_bst else None,
                "evals_result": evals_result,
                "train_n": int(dm.n_rows),
            }
            self.dist_callbacks.after_train(self, result)
            return result
        finally:
            coll.shutdown()

    def predict(self, model_bytes: bytes, data_uid: str, kwargs: Dict) -> np.ndarray:
        import torch

        self.dist_callbacks.before_predict(self)
        bst = pickle.loads(model_bytes) if isinstance(
            model_bytes, (bytes, bytearray)
        ) else model_bytes
        shard = self.shards.get(data_uid)
        if shard is None:
            raise RuntimeError(f"Prediction data {data_uid} not loaded")
        it_range = kwargs.get("iteration_range")
        if kwargs.get("pred_leaf") or kwargs.get("pred_contribs"):
            # leaf-index / SHAP modes run per shard on the CPU tree walk
            if shard.get("streaming"):
                parts = [
                    bst.predict(chunk["data"], **kwargs)
                    for chunk in shard["loader"].iter_shards(
                        shard["rank"], shard["num_actors"]
                    )
                ]
                out = np.concatenate(parts, axis=0)
            else:
                out = bst.predict(shard["data"], **kwargs)
            self.dist_callbacks.after_predict(self, out)
            return out
        if shard.get("streaming"):
            loader = shard["loader"]
            margins = []
            for chunk in loader.iter_shards(shard["rank"], shard["num_actors"]):
                Xc = torch.from_numpy(
                    np.ascontiguousarray(chunk["data"], np.float32)
                ).to(self.device)
                mg = bst.predict_margin_tensor(Xc, it_range)
                bm = chunk.get("base_margin")
                if bm is not None:
                    mg = mg + torch.from_numpy(
                        np.ascontiguousarray(bm, np.float32)
                    ).to(mg.device).reshape(
                        mg.shape[0], *([1] * (mg.dim() - 1))
                    )
                margins.append(mg)
                del Xc
            margin = torch.cat(margins, dim=0)
        else:
            X = torch.from_numpy(
                np.ascontiguousarray(shard["data"], np.float32)
            ).to(self.device)
            margin = bst.predict_margin_tensor(X, it_range)
            # xgboost semantics: a base_margin on the prediction matrix
            # is added to the raw margin before any transform
            bm = shard.get("base_margin")
            if bm is not None:
                margin = margin + torch.from_numpy(
                    np.ascontiguousarray(bm, np.float32)
                ).to(margin.device).reshape(
                    margin.shape[0], *([1] * (margin.dim() - 1))
                )
        if kwargs.get("output_margin"):
            pred = margin.cpu().numpy()
        else:
            from xgboost_ray_amd.engine.objectives import get_objective

            obj = get_objective(bst.objective, bst.num_class)
            pred = obj.transform_prediction(margin).cpu().numpy()
        self.dist_callbacks.after_predict(self, pred)
        return pred

    def unload_data(self, uids: Optional[List[str]] = None):
        if uids is None:
            self.shards = {}
        else:
            for uid in uids:
                self.shards.pop(uid, None)
        return True

    def shutdown(self):
        return "__shutdown__"


class _StopCallback:
    """Cooperative stop: checked after every boosting round
    (reference _StopCallback, main.py:628-652)."""

    def __init__(self, stop_event):
        self.stop_event = stop_event
        self.stopped = False

    def after_iteration(self, booster, iteration, evals_log) -> bool:
        if self.stop_event is not None and self.stop_event.is_set():
            self.stopped = True
            return True
        return False


class _CheckpointCallback:
    """Rank-0 checkpointing through the queue
    (reference _SaveInternalCheckpointCallback, main.py:612-626)."""

    def __init__(self, frequency: int, queue, rank: int):
        self.frequency = frequency
        self.queue = queue
        self.rank = rank
        self._within = -1

    def after_iteration(self, booster, iteration, evals_log) -> bool:
        self._within += 1
        if self.frequency and (self._within + 1) % self.frequency == 0:
            self.queue.put(
                (self.rank, ("__checkpoint__", self._within, pickle.dumps(booster)))
            )
        return False


def _to_dev(arr, device, dtype=np.float32):
    import torch

    if arr is None:
        return None
    if dtype is not None:
        arr = np.ascontiguousarray(arr, dtype=dtype)
    else:
        arr = np.ascontiguousarray(arr)
        if arr.dtype == object or not np.issubdtype(arr.dtype, np.number):
            # qid as strings etc: factorize to ints
            _, arr = np.unique(arr, return_inverse=True)
        arr = arr.astype(np.int64) if arr.dtype.kind in "iu" else arr
    return torch.from_numpy(arr).to(device)


def _materialize_shard(desc: dict) -> Dict[str, Any]:
    from xgboost_ray_amd import shm_store

    if desc["kind"] == "refs":
        shard = {}
        for key, val in desc["refs"].items():
            if isinstance(val, shm_store.ObjectRef):
                shard[key] = shm_store.get(val)
            else:
                shard[key] = val
        return shard
    if desc["kind"] == "distributed":
        loader = desc["loader"]
        shard, _ = loader.load_data(
            desc["num_actors"], None, rank=desc["rank"]
        )
        return shard
    if desc["kind"] == "streaming":
        # defer materialization: the training path streams file-by-file
        # through the device sketch + binning kernels
        return {
            "streaming": True,
            "loader": desc["loader"],
            "rank": desc["rank"],
            "num_actors": desc["num_actors"],
            "data": None,
        }
    if desc["kind"] == "inline":
        return desc["shard"]
    raise ValueError(f"Unknown shard descriptor kind: {desc['kind']}")


def _actor_main(rank, world_size, conn, queue, stop_event, config):
    """Entry point of the actor process."""
    import warnings as _warnings

    # shared-memory shards are mapped read-only; torch zero-copy views of
    # them are never written, so the non-writable warning is noise
    _warnings.filterwarnings("ignore", message=".*not writable.*")
    worker = None
    try:
        worker = _ActorWorker(rank, world_size, queue, stop_event, config)
    except Exception:
        conn.send(("__init__", "err", traceback.format_exc()))
        return
    conn.send(("__init__", "ok", os.getpid()))
    while True:
        try:
            msg = conn.recv()
        except (EOFError, KeyboardInterrupt):
            break
        call_id, method, args, kwargs = msg
        try:
            fn = getattr(worker, method)
            result = fn(*args, **kwargs)
            conn.send((call_id, "ok", result))
            if result == "__shutdown__":
                break
        except TrainingStoppedError as e:
            conn.send((call_id, "stopped", str(e)))
        except Exception:
            conn.send((call_id, "err", traceback.format_exc()))


# --------------------------------------------------------------------------
# Driver side
# --------------------------------------------------------------------------


class ActorHandle:
    """Driver-side handle with Ray-like ``.remote()`` call semantics."""

    def __init__(
        self,
        rank: int,
        world_size: int,
        queue,
        stop_event,
        config: Optional[dict] = None,
        gpu_id: Optional[int] = None,
    ):
        self.rank = rank
        self.world_size = world_size
        self.queue = queue
        self.stop_event = stop_event
        self.config = dict(config or {})
        self.gpu_id = gpu_id
        self._proc: Optional[mp.Process] = None
        self._conn = None
        self._pid = None
        self._futures: Dict[int, Any] = {}
        self._next_id = 0
        self._lock = threading.Lock()
        self._reader: Optional[threading.Thread] = None
        self._dead = False

    def start(self, timeout: Optional[float] = None):
        from xgboost_ray_amd.util import Future

        ctx = mp.get_context("spawn")
        parent_conn, child_conn = ctx.Pipe()
        env_backup = {}
        try:
            if self.gpu_id is not None:
                for var in ("HIP_VISIBLE_DEVICES", "CUDA_VISIBLE_DEVICES"):
                    env_backup[var] = os.environ.get(var)
                    os.environ[var] = str(self.gpu_id)
            os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
            self._proc = ctx.Process(
                target=_actor_main,
                args=(
                    self.rank,
                    self.world_size,
                    child_conn,
                    self.queue,
                    self.stop_event,
                    self.config,
                ),
                daemon=True,
            )
            self._proc.start()
        finally:
            for var, val in env_backup.items():
                if val is None:
                    os.environ.pop(var, None)
                else:
                    os.environ[var] = val
        child_conn.close()
        self._conn = parent_conn
        # wait for init handshake
        deadline = time.monotonic() + (timeout or ENV.ACTOR_START_TIMEOUT_S)
        while not self._conn.poll(0.1):
            if time.monotonic() > deadline:
                self.kill()
                raise ActorError(f"Actor {self.rank} failed to start in time")
            if not self._proc.is_alive():
                raise ActorError(f"Actor {self.rank} died during startup")
        tag, status, payload = self._conn.recv()
        if status != "ok":
            raise ActorError(f"Actor {self.rank} init failed:\n{payload}")
        self._pid = payload
        self._reader = threading.Thread(target=self._read_loop, daemon=True)
        self._reader.start()
        return self

    def _read_loop(self):
        while True:
            try:
                call_id, status, payload = self._conn.recv()
            except (EOFError, OSError):
                self._on_death()
                return
            with self._lock:
                fut = self._futures.pop(call_id, None)
            if fut is None:
                continue
            if status == "ok":
                fut.set_result(payload)
            elif status == "stopped":
                fut.set_error(TrainingStoppedError(payload))
            else:
                fut.set_error(TrainingError(f"Actor {self.rank}:\n{payload}"))

    def _on_death(self):
        self._dead = True
        with self._lock:
            futs = list(self._futures.values())
            self._futures = {}
        for fut in futs:
            fut.set_error(
                ActorError(f"Actor {self.rank} (pid={self._pid}) died")
            )

    def remote(self, method: str, *args, **kwargs):
        from xgboost_ray_amd.util import Future

        fut = Future(actor=self, method=method)
        if self._dead or self._conn is None:
            fut.set_error(ActorError(f"Actor {self.rank} is not alive"))
            return fut
        with self._lock:
            call_id = self._next_id
            self._next_id += 1
            self._futures[call_id] = fut
        try:
            self._conn.send((call_id, method, args, kwargs))
        except (BrokenPipeError, OSError):
            self._on_death()
        return fut

    def is_alive(self) -> bool:
        return (
            not self._dead
            and self._proc is not None
            and self._proc.is_alive()
        )

    def pid(self):
        return self._pid

    def ip(self):
        return "127.0.0.1"

    def kill(self):
        self._dead = True
        if self._proc is not None and self._proc.is_alive():
            self._proc.terminate()
            self._proc.join(timeout=5)
            if self._proc.is_alive():
                self._proc.kill()
                self._proc.join(timeout=5)

    def shutdown(self, graceful: bool = True):
        if graceful and self.is_alive():
            try:
                fut = self.remote("shutdown")
                fut.result(timeout=ENV.ACTOR_SHUTDOWN_TIMEOUT_S)
            except Exception:
                pass
        if self._proc is not None:
            self._proc.join(timeout=ENV.ACTOR_SHUTDOWN_TIMEOUT_S)
        self.kill()
