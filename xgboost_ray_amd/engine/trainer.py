"""The boosting engine: depth-wise GPU histogram tree growing.

MI355X-native replacement for XGBoost's ``gpu_hist`` updater + training
loop (the reference delegates this entirely to ``xgb.train``, reference
main.py:745; SURVEY.md #2.3 row 2 calls it the centerpiece). Per round:

  grad/hess kernel -> int64 fixed-point quantization -> per-depth
  LDS-tiled histogram build (smaller child only; sibling derived by the
  histogram-subtraction trick) -> RCCL AllReduce of the built histograms
  over xGMI -> best-split scan -> stable row partition -> next depth.

Int64 fixed-point gradients make every reduction an integer sum:
bitwise-deterministic regardless of GPU atomic ordering, feature-block
schedule, or world size - which is what makes checkpoint-resume models
exactly reproducible (reference test_fault_tolerance.py:401-449).
"""

import math
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional, Sequence, Tuple, Union

import numpy as np
import torch

from xgboost_ray_amd import ops
from xgboost_ray_amd.booster import Booster, Tree
from xgboost_ray_amd.engine.collective import Collective
from xgboost_ray_amd.engine.metrics import get_metric
from xgboost_ray_amd.engine.objectives import Objective, get_objective
from xgboost_ray_amd.engine.quantile import BinnedMatrix

_QUANT_BITS = 30


@dataclass
class TrainParams:
    objective: Union[str, Callable] = "reg:squarederror"
    eta: float = 0.3
    max_depth: int = 6
    max_bin: int = 256
    reg_lambda: float = 1.0
    reg_alpha: float = 0.0
    gamma: float = 0.0
    min_child_weight: float = 1.0
    max_delta_step: float = 0.0
    subsample: float = 1.0
    num_parallel_tree: int = 1
    colsample_bytree: float = 1.0
    colsample_bylevel: float = 1.0
    colsample_bynode: float = 1.0
    scale_pos_weight: float = 1.0
    base_score: Optional[float] = None
    num_class: int = 0
    eval_metric: List[str] = field(default_factory=list)
    seed: int = 0
    grow_policy: str = "depthwise"
    max_leaves: int = 0
    monotone_constraints: object = None
    interaction_constraints: object = None
    tweedie_variance_power: float = 1.5
    aft_loss_distribution: str = "normal"
    aft_loss_distribution_scale: float = 1.0
    tree_method: str = "hist"
    booster: str = "gbtree"
    # DART (booster="dart") dropout parameters (xgboost semantics)
    rate_drop: float = 0.0
    skip_drop: float = 0.0
    one_drop: bool = False
    normalize_type: str = "tree"
    sample_type: str = "uniform"  # or "weighted" (xgboost dart.cc)
    verbosity: int = 1
    nthread: int = 0
    disable_default_eval_metric: bool = False

    @classmethod
    def from_dict(cls, params: Dict) -> "TrainParams":
        p = cls()
        alias = {
            "learning_rate": "eta",
            "lambda": "reg_lambda",
            "alpha": "reg_alpha",
            "min_split_loss": "gamma",
            "random_state": "seed",
            "n_jobs": "nthread",
        }
        for k, v in (params or {}).items():
            k = alias.get(k, k)
            if k == "eval_metric":
                p.eval_metric = [v] if isinstance(v, str) else list(v)
            elif hasattr(p, k):
                setattr(p, k, v)
            # unknown params are accepted silently (xgboost behavior)
        p.eta = float(p.eta)
        p.max_depth = int(p.max_depth)
        p.max_bin = int(p.max_bin)
        p.num_class = int(p.num_class or 0)
        p.seed = int(p.seed or 0)
        p.num_parallel_tree = max(1, int(p.num_parallel_tree or 1))
        if isinstance(p.monotone_constraints, str):
            txt = p.monotone_constraints.strip().strip("()")
            p.monotone_constraints = [
                int(x) for x in txt.split(",") if x.strip()
            ]
        if isinstance(p.interaction_constraints, str):
            import ast

            p.interaction_constraints = ast.literal_eval(
                p.interaction_constraints
            )
        return p


@dataclass
class EvalPack:
    """One eval set: raw features + incremental margin state."""

    name: str
    X: Optional[torch.Tensor]  # None => the training matrix itself
    label: Optional[torch.Tensor] = None
    weight: Optional[torch.Tensor] = None
    qid: Optional[torch.Tensor] = None
    base_margin: Optional[torch.Tensor] = None
    margin: Optional[torch.Tensor] = None


@dataclass(slots=True)
class _Node:
    nid: int
    depth: int
    start: int
    count: int
    sum_g: int  # quantized
    sum_h: int
    slot: int = 0  # scan-slot in this depth's all_hist
    parent_slot: int = 0  # parent's scan-slot in the previous depth
    w_lower: float = -math.inf  # monotone weight bounds
    w_upper: float = math.inf
    path_feats: tuple = ()  # split features on the root path (sorted)


class _TreeArrays:
    """Growable SoA node storage for depthwise growth (replaces python
    lists of per-node appends; fields are written by vectorized numpy
    fancy-indexing per depth)."""

    __slots__ = ("feat", "thr", "left", "dl", "val", "gain", "cover",
                 "parent", "n", "cap")
    _FIELDS = (
        ("feat", -1, np.int64), ("thr", 0.0, np.float64),
        ("left", -1, np.int64), ("dl", 0, np.int64),
        ("val", 0.0, np.float64), ("gain", 0.0, np.float64),
        ("cover", 0.0, np.float64), ("parent", -1, np.int64),
    )

    def __init__(self, cap: int = 64):
        self.cap = int(cap)
        self.n = 1  # root preallocated (all defaults)
        for name, fill, dt in self._FIELDS:
            setattr(self, name, np.full(self.cap, fill, dt))

    def ensure(self, need: int):
        if need <= self.cap:
            return
        cap = self.cap
        while cap < need:
            cap *= 2
        for name, fill, dt in self._FIELDS:
            old = getattr(self, name)
            new = np.full(cap, fill, dt)
            new[: self.n] = old[: self.n]
            setattr(self, name, new)
        self.cap = cap


def _calc_weight_vec(G, H, lam, alpha):
    """Vectorized _calc_weight with the identical operation order (the
    scalar path's results are part of the bitwise determinism contract)."""
    denom = H + lam
    if alpha > 0:
        G = np.copysign(np.maximum(np.abs(G) - alpha, 0.0), G)
    bad = denom <= 0
    return np.where(bad, 0.0, -G / np.where(bad, 1.0, denom))


class CallbackList:
    def __init__(self, callbacks):
        self.callbacks = list(callbacks or [])

    def before_iteration(self, booster, iteration, evals_log) -> bool:
        stop = False
        for cb in self.callbacks:
            if hasattr(cb, "before_iteration"):
                stop |= bool(cb.before_iteration(booster, iteration, evals_log))
        return stop

    def after_iteration(self, booster, iteration, evals_log) -> bool:
        stop = False
        for cb in self.callbacks:
            if hasattr(cb, "after_iteration"):
                stop |= bool(cb.after_iteration(booster, iteration, evals_log))
            elif callable(cb):
                stop |= bool(cb(booster, iteration, evals_log))
        return stop

    def after_training(self, booster, iteration, evals_log):
        """End-of-training hook. Accepts both this engine's 3-arg form
        (booster, final_iteration, log) and xgboost's 1-arg
        TrainingCallback.after_training(model) signature."""
        import inspect

        for cb in self.callbacks:
            fn = getattr(cb, "after_training", None)
            if fn is None:
                continue
            try:
                n_params = len(inspect.signature(fn).parameters)
            except (TypeError, ValueError):
                n_params = 3
            if n_params >= 3:
                fn(booster, iteration, evals_log)
            else:
                fn(booster)


class BoostingEngine:
    """Single-rank engine; collectives make it data-parallel."""

    def __init__(
        self,
        params: Dict,
        dtrain: BinnedMatrix,
        collective: Optional[Collective] = None,
        rank: int = 0,
        custom_objective: Optional[Callable] = None,
    ):
        self.p = TrainParams.from_dict(params)
        self.raw_params = dict(params or {})
        self.dtrain = dtrain
        self.coll = collective or Collective()
        self.rank = rank
        self.device = dtrain.bins.device
        # pinned double-use staging buffers for tiny per-depth H2D pushes
        # (pageable copies block the host in the driver's staging path);
        # each slot is drained by a stream-ordered D2H pull before reuse
        self._pin_bufs = {}
        obj_spec = custom_objective or self.p.objective
        self.obj: Objective = get_objective(
            obj_spec, self.p.num_class, float(self.p.scale_pos_weight),
            float(self.p.tweedie_variance_power),
        )
        # xgboost defaults max_delta_step to 0.7 for Poisson regression
        if (self.p.objective == "count:poisson"
                and not self.p.max_delta_step):
            self.p.max_delta_step = 0.7
        from xgboost_ray_amd.engine.objectives import AFT

        if isinstance(self.obj, AFT):
            self.obj.dist = str(self.p.aft_loss_distribution)
            self.obj.sigma = float(self.p.aft_loss_distribution_scale)
        self.n_class = max(1, self.p.num_class)
        if self.p.base_score is None:
            self.p.base_score = 0.5
        self.base_margin_const = self.obj.prob_to_margin(self.p.base_score)

        n = dtrain.n_rows
        shape = (n,) if self.n_class == 1 else (n, self.n_class)
        self.margin = torch.full(
            shape, self.base_margin_const, dtype=torch.float32, device=self.device
        )
        if dtrain.base_margin is not None:
            bm = dtrain.base_margin.reshape(shape)
            self.margin = self.margin + bm
        if dtrain.label is not None:
            self.obj.validate_label(dtrain.label)
        self._train_label = dtrain.label
        if isinstance(self.obj, AFT):
            yl = getattr(dtrain, "label_lower_bound", None)
            yu = getattr(dtrain, "label_upper_bound", None)
            if yl is None or yu is None:
                if dtrain.label is None:
                    raise ValueError(
                        "survival:aft needs label_lower_bound/"
                        "label_upper_bound (or a plain label for the "
                        "uncensored case)"
                    )
                yl = yu = dtrain.label
            self._train_label = torch.stack(
                [yl.float(), yu.float()], dim=1
            )
        self.feat_bins = dtrain.cuts.feat_bins().to(self.device)
        self.n_bins = dtrain.cuts.max_bins
        self.mono = None
        mc = self.p.monotone_constraints
        if mc:
            arr = np.zeros(dtrain.n_features, dtype=np.int8)
            for i, c in enumerate(list(mc)[: dtrain.n_features]):
                arr[i] = int(c)
            if arr.any():
                self.mono = torch.from_numpy(arr).to(self.device)
                if self.p.grow_policy == "lossguide":
                    raise ValueError(
                        "monotone_constraints are not supported with "
                        "grow_policy=lossguide yet"
                    )
        # interaction constraints: list of feature-index groups
        # (reference defers to xgboost's `interaction_constraints`;
        # semantics: a split feature must share a constraint set with
        # every feature already used on the node's root path)
        self.interaction_sets = None
        ic = self.p.interaction_constraints
        if ic:
            self.interaction_sets = [frozenset(int(f) for f in g)
                                     for g in ic]
        self.lin_w = None
        if self.p.booster == "gblinear":
            raw = getattr(dtrain, "raw_X", None)
            if raw is None:
                raise ValueError(
                    "booster=gblinear needs the raw feature matrix - "
                    "build the BinnedMatrix with keep_raw=True (streaming "
                    "matrices are not supported for gblinear)"
                )
            self.lin_w = torch.zeros(
                (dtrain.n_features + 1, self.n_class),
                dtype=torch.float64, device=self.device,
            )
        elif self.p.booster not in ("gbtree", "dart", ""):
            raise ValueError(
                f"Unsupported booster: {self.p.booster!r} "
                "(gbtree/dart/gblinear)"
            )
        self.iteration = 0
        self.booster = Booster(
            params={
                "objective": getattr(self.obj, "name", "custom"),
                "base_score": self.p.base_score,
                "num_class": self.p.num_class,
                "num_feature": dtrain.n_features,
                "max_depth": self.p.max_depth,
                "eta": self.p.eta,
                "num_parallel_tree": self.p.num_parallel_tree,
                # retained so predict() can route to GPU actors
                # (_is_gpu_params reads the model's tree_method)
                "tree_method": self.p.tree_method,
                "booster": self.p.booster,
            }
        )

    # -- resume ------------------------------------------------------------
    def load_model(self, model: Booster):
        """Continue training from an existing model (xgb_model kwarg)."""
        self.booster = model
        self.iteration = model.num_boosted_rounds()
        # restore the dart weight ledger (weighted dropout resume)
        dw = model.attr("dart_weights")
        if dw:
            self._dart_w = [float(x) for x in dw.split(",") if x]
        # recompute margins through the binned matrix (thresholds are exact
        # cut values for models we trained, so binned replay is exact)
        n = self.dtrain.n_rows
        shape = (n,) if self.n_class == 1 else (n, self.n_class)
        base = self.obj.prob_to_margin(model.base_score)
        self.base_margin_const = base
        self.margin = torch.full(
            shape, base, dtype=torch.float32, device=self.device
        )
        if self.dtrain.base_margin is not None:
            self.margin = self.margin + self.dtrain.base_margin.reshape(shape)
        if getattr(model, "linear_weights", None) is not None:
            # gblinear resume: restore weights, recompute margins exactly
            self.lin_w = torch.from_numpy(
                np.asarray(model.linear_weights, np.float64)
            ).to(self.device).reshape(self.dtrain.n_features + 1,
                                      self.n_class)
            X = self.dtrain.raw_X
            w32 = self.lin_w.float()
            m = X @ w32[:-1] + w32[-1]
            self.margin = (
                self.margin + (m[:, 0] if self.n_class == 1 else m)
            )
            return
        self._replay_trees_binned(model)

    def _replay_trees_binned(self, model: Booster):
        cuts = self.dtrain.cuts
        cut_ptr = cuts.cut_ptr.cpu().numpy()
        cuts_np = cuts.cuts_flat.cpu().numpy()
        for i, t in enumerate(model.trees):
            cls = model.tree_info[i] if i < len(model.tree_info) else 0
            split_bin = np.zeros(t.num_nodes, dtype=np.int32)
            for nid in range(t.num_nodes):
                if t.feat[nid] >= 0:
                    f = int(t.feat[nid])
                    lo, hi = cut_ptr[f], cut_ptr[f + 1]
                    # stored thr == cut[b] for a "bin <= b" split, so the
                    # bin index is exactly the position of thr in the cuts
                    b = np.searchsorted(cuts_np[lo:hi], t.thr[nid], side="left")
                    split_bin[nid] = b
            margin_view = (
                self.margin if self.n_class == 1 else self.margin[:, cls]
            )
            self._add_tree_margin_binned(t, split_bin, margin_view)

    def _add_tree_margin_binned(self, t: Tree, split_bin, margin_view):
        bins = self.dtrain.bins
        n = bins.shape[0]
        cur = torch.zeros(n, dtype=torch.int64, device=self.device)
        feat = torch.from_numpy(t.feat.astype(np.int64)).to(self.device)
        sbin = torch.as_tensor(split_bin, dtype=torch.int64, device=self.device)
        left = torch.from_numpy(t.left.astype(np.int64)).to(self.device)
        dl = torch.from_numpy(t.default_left.astype(np.uint8)).to(self.device)
        val = torch.from_numpy(t.value).to(self.device)
        while True:
            f = feat[cur]
            active = f >= 0
            if not bool(active.any()):
                break
            rows = torch.nonzero(active).flatten()
            fa = f[rows]
            bv = bins[rows, fa].long()
            miss = bv == ops.MISSING_BIN
            go_left = bv <= sbin[cur[rows]]
            go_left = torch.where(miss, dl[cur[rows]].bool(), go_left)
            cur[rows] = torch.where(go_left, left[cur[rows]], left[cur[rows]] + 1)
        margin_view += val[cur]

    # -- one boosting round --------------------------------------------------
    def update(self):
        it = self.iteration
        label = self._train_label
        dart_ctx = None
        if self.p.booster == "dart":
            dart_ctx = self._dart_dropout(it)
        gpair = self.obj.gradients(
            self.margin, label, self.dtrain.weight, self.dtrain.qid
        )
        if self.p.booster == "gblinear":
            self._update_linear(gpair)
            self.iteration += 1
            return []
        trees, classes = [], []
        for cls in range(self.n_class):
            gp = gpair if self.n_class == 1 else gpair[:, cls, :]
            # num_parallel_tree > 1: boosted random forest - k trees per
            # round on the same gradients with independent row/column
            # samples (XGBoost GBTree::BoostNewTrees semantics)
            for ptree in range(self.p.num_parallel_tree):
                tree = self._grow_tree(
                    gp.contiguous(), it, cls, ptree=ptree
                )
                trees.append(tree)
                classes.append(cls)
        self.booster.append_round(trees, classes)
        if dart_ctx is not None:
            self._dart_commit(dart_ctx, trees, classes)
        self.iteration += 1
        return trees


    def _update_linear(self, gpair):
        """One gblinear round: preconditioned full-batch coordinate step.

        The reference reaches gblinear through xgboost's booster param
        passthrough (reference main.py:745 xgb.train(params...); exercised
        by reference tests/test_sklearn.py test_sklearn_api_gblinear).

        Deterministic replacement for xgboost's gblinear updaters
        (shotgun is non-deterministic; this is a damped Jacobi update
        with the same per-coordinate Newton preconditioner, L2 on
        weights and L1 soft-thresholding, eta-damped). Distributed: one
        fp64 AllReduce of [X^T g | (X*X)^T h | sum_g | sum_h] per class.
        """
        X = self.dtrain.raw_X
        n, F = X.shape
        lam = float(self.p.reg_lambda)
        alpha = float(self.p.reg_alpha)
        eta = float(self.p.eta)
        deltas = []
        for cls in range(self.n_class):
            gp = gpair if self.n_class == 1 else gpair[:, cls, :]
            # chunked fp64 accumulation: fixed chunk order -> identical
            # sums on every rank layout of the same data
            sg = torch.zeros(F, dtype=torch.float64, device=self.device)
            sh = torch.zeros(F, dtype=torch.float64, device=self.device)
            sb = torch.zeros(2, dtype=torch.float64, device=self.device)
            CH = 1 << 20
            for s0 in range(0, n, CH):
                Xc = X[s0 : s0 + CH].double()
                gc = gp[s0 : s0 + CH, 0].double()
                hc = gp[s0 : s0 + CH, 1].double()
                sg += Xc.t() @ gc
                sh += (Xc * Xc).t() @ hc
                sb[0] += gc.sum()
                sb[1] += hc.sum()
            if self.coll.is_distributed:
                packed = torch.cat([sg, sh, sb])
                self.coll.allreduce_(packed)
                sg, sh, sb = packed[:F], packed[F : 2 * F], packed[2 * F :]
            w = self.lin_w[:F, cls]
            denom = sh + lam
            full = w - (sg + lam * w) / denom
            if alpha > 0.0:
                thr = alpha / denom
                full = torch.sign(full) * torch.clamp(
                    full.abs() - thr, min=0.0
                )
            dw = eta * (full - w)
            db = eta * (-sb[0] / (sb[1] + lam))
            self.lin_w[:F, cls] += dw
            self.lin_w[F, cls] += db
            dwf = dw.float()
            mv = self.margin if self.n_class == 1 else self.margin[:, cls]
            for s0 in range(0, n, CH):
                mv[s0 : s0 + CH] += X[s0 : s0 + CH] @ dwf
            mv += float(db)
            deltas.append((dwf, float(db)))
        self._lin_deltas = deltas
        self.booster.linear_weights = self.lin_w.cpu().numpy()
        self.booster.linear_rounds = self.iteration + 1
        self.booster.params["booster"] = "gblinear"


    # -- DART (dropout) ----------------------------------------------------
    def _dart_predict_trees(self, tree_ids):
        """Margin contribution of the given trees on the training rows."""
        contrib = torch.zeros_like(self.margin)
        for ti in tree_ids:
            t = self.booster.trees[ti]
            cls = self.booster.tree_info[ti]
            mv = contrib if self.n_class == 1 else contrib[:, cls]
            self._add_tree_margin_raw(t, mv)
        return contrib

    def _add_tree_margin_raw(self, t, margin_view):
        """Add one tree's predictions via the binned matrix (exact)."""
        cuts = self.dtrain.cuts
        cut_ptr = cuts.cut_ptr.cpu().numpy()
        cuts_np = cuts.cuts_flat.cpu().numpy()
        split_bin = np.zeros(t.num_nodes, dtype=np.int32)
        for nid in range(t.num_nodes):
            if t.feat[nid] >= 0:
                f = int(t.feat[nid])
                lo, hi = cut_ptr[f], cut_ptr[f + 1]
                split_bin[nid] = np.searchsorted(
                    cuts_np[lo:hi], t.thr[nid], side="left"
                )
        self._add_tree_margin_binned(t, split_bin, margin_view)

    def _dart_dropout(self, it):
        """Pick the dropped-tree set and remove it from the margins.

        The reference reaches DART through xgboost's booster="dart"
        passthrough (reference main.py:745); semantics follow xgboost
        dart.cc (rate_drop/skip_drop/one_drop/normalize_type).

        Deterministic: seeded by (seed, iteration), identical on every
        rank. Returns (dropped_ids, dropped_contrib) or a no-drop marker.
        """
        n_trees = len(self.booster.trees)
        rng = np.random.RandomState(
            (int(self.p.seed) * 77003 + it * 9973) % (2**31)
        )
        if n_trees == 0 or rng.rand() < float(self.p.skip_drop):
            return ([], None)
        if self.p.sample_type == "weighted":
            # xgboost dart.cc weighted selection: drop probability
            # proportional to the tree's current weight,
            # p_i = w_i * rate_drop * n / sum(w), clamped to 1
            w = np.asarray(self._dart_weights(n_trees), dtype=np.float64)
            sw = float(w.sum())
            if sw <= 0:
                p_i = np.full(n_trees, float(self.p.rate_drop))
            else:
                p_i = np.clip(
                    w * float(self.p.rate_drop) * n_trees / sw, 0.0, 1.0
                )
            drop = np.nonzero(rng.rand(n_trees) < p_i)[0]
        else:
            drop = np.nonzero(
                rng.rand(n_trees) < float(self.p.rate_drop)
            )[0]
        if drop.size == 0 and self.p.one_drop:
            drop = np.array([int(rng.randint(n_trees))])
        if drop.size == 0:
            return ([], None)
        dropped = [int(d) for d in drop]
        contrib = self._dart_predict_trees(dropped)
        self.margin -= contrib
        return (dropped, contrib)

    def _dart_weights(self, n_trees):
        """Per-tree weights for weighted dropout. Scales are baked into
        leaf values at commit time, so the weight ledger is maintained
        alongside (new trees enter at their new_scale; dropped trees are
        rescaled by old_scale; untouched trees keep weight 1 from their
        no-drop round)."""
        w = getattr(self, "_dart_w", None)
        if w is None:
            w = []
            self._dart_w = w
        while len(w) < n_trees:
            w.append(1.0)
        return w  # the LIVE ledger (callers mutate it in place)

    def _dart_commit(self, dart_ctx, trees, classes):
        """Scale the new + dropped trees (xgboost normalize_type) and
        restore margins. Scales are baked into leaf values so predict
        and serialization need no per-tree weight array."""
        dropped, contrib = dart_ctx
        k = len(dropped)
        if k == 0:
            return
        # xgboost dart.cc NormalizeTrees: leaves already carry eta, the
        # weights below redistribute mass between new and dropped trees
        lr = float(self.p.eta) / max(1, len(trees) // self.n_class)
        if self.p.normalize_type == "forest":
            new_scale = 1.0 / (1.0 + lr)
            old_scale = 1.0 / (1.0 + lr)
        else:  # "tree"
            new_scale = 1.0 / (k + lr)
            old_scale = k / (k + lr)
        # scale the just-built trees' leaves (and their margin effect,
        # which update() already added at full weight)
        delta = torch.zeros_like(self.margin)
        for t, cls in zip(trees, classes):
            mv = delta if self.n_class == 1 else delta[:, cls]
            self._add_tree_margin_raw(t, mv)
            t.value *= new_scale
        self.margin += (new_scale - 1.0) * delta
        self.booster._flat_cache = None
        # scale the dropped trees and re-add their shrunk contribution
        for ti in dropped:
            self.booster.trees[ti].value *= old_scale
        self.margin += old_scale * contrib
        # weight ledger for sample_type="weighted"; persisted as a
        # booster attribute so checkpoint/resume reproduces the same
        # weighted-dropout probabilities (fuzz-found)
        wl = self._dart_weights(len(self.booster.trees))
        for ti in dropped:
            wl[ti] *= old_scale
        for i in range(len(trees)):
            wl[len(self.booster.trees) - len(trees) + i] = new_scale
        self.booster.set_attr(
            dart_weights=",".join(repr(w) for w in wl)
        )

    def _quantize(self, gpair: torch.Tensor) -> Tuple[torch.Tensor, float, float]:
        fused_mx = getattr(gpair, "_rxgb_absmax", None)
        if fused_mx is not None:
            # grad_fused already reduced |g|/|h| maxes in its single pass
            # (f32 max is order-free, so the value equals abs().max())
            mx = fused_mx.double()
        else:
            mx = torch.stack(
                [gpair[:, 0].abs().max(), gpair[:, 1].abs().max()]
            ).double()
        if self.coll.is_distributed:
            mx_d = mx.to(self.device) if self.device.type == "cuda" else mx
            self.coll.allreduce_(mx_d, op="max")
            mx = mx_d.cpu()
        scale_g = float(2.0**_QUANT_BITS) / max(float(mx[0]), 1e-300)
        scale_h = float(2.0**_QUANT_BITS) / max(float(mx[1]), 1e-300)
        return ops.quantize_gpair(gpair, scale_g, scale_h), scale_g, scale_h

    def _sample_rows(self, it: int, cls: int) -> torch.Tensor:
        n = self.dtrain.n_rows
        if self.p.subsample >= 1.0:
            return torch.arange(n, dtype=torch.int32, device=self.device)
        gen = torch.Generator(device="cpu")
        gen.manual_seed(
            (self.p.seed * 2654435761 + it * 97 + cls * 31 + self.rank) % (2**63)
        )
        mask = torch.rand(n, generator=gen) < self.p.subsample
        return torch.nonzero(mask.to(self.device)).flatten().to(torch.int32)

    def _sample_features(
        self, it: int, cls: int, level: Optional[int] = None
    ) -> Optional[torch.Tensor]:
        """Feature mask for this tree (level=None) or depth level.

        Seeded identically on every rank (global seed, no rank term) so
        all workers mask the same features. feature_weights bias the
        selection (xgboost semantics)."""
        F = self.dtrain.n_features
        frac = (
            self.p.colsample_bytree if level is None else self.p.colsample_bylevel
        )
        if frac >= 1.0:
            return None
        k = max(1, int(round(F * frac)))
        gen = torch.Generator(device="cpu")
        salt = 7 if level is None else 1009 + level
        gen.manual_seed(
            (self.p.seed * 2654435761 + it * 97 + cls * 31 + salt) % (2**63)
        )
        fw = getattr(self.dtrain, "feature_weights", None)
        if fw is not None:
            w = fw.cpu().double().clamp(min=0)
            if float(w.sum()) > 0:
                perm = torch.multinomial(
                    w, k, replacement=False, generator=gen
                )
            else:
                perm = torch.randperm(F, generator=gen)[:k]
        else:
            perm = torch.randperm(F, generator=gen)[:k]
        mask = torch.zeros(F, dtype=torch.bool)
        mask[perm] = True
        return mask.to(self.device)


    def _stage_i64(self, arr, slot):
        """Pinned H2D staging for a small int64 numpy vector (GPU only)."""
        arr = np.ascontiguousarray(arr, dtype=np.int64).reshape(-1)
        n = arr.size
        buf = self._pin_bufs.get(slot)
        if buf is None or buf.numel() < n:
            cap = max(4096, 1 << max(0, (n - 1)).bit_length())
            buf = torch.empty(cap, dtype=torch.int64, pin_memory=True)
            self._pin_bufs[slot] = buf
        view = buf[:n]
        view.copy_(torch.from_numpy(arr))
        return view.to(self.device, non_blocking=True)


    def _bynode_mask(self, KK, it, cls, depth):
        """[KK, F] uint8 per-node feature gate for colsample_bynode.
        Multiplies with colsample_bytree/bylevel (those zero feat_bins)
        and interaction constraints (ANDed by the caller)."""
        F = self.dtrain.n_features
        k = max(1, int(round(float(self.p.colsample_bynode) * F)))
        rng = np.random.RandomState(
            (self.p.seed * 48271 + it * 2246822519
             + cls * 3266489917 + depth * 668265263) % (2**31)
        )
        m = np.zeros((KK, F), np.uint8)
        for row in range(KK):
            m[row, rng.choice(F, size=k, replace=False)] = 1
        return torch.from_numpy(m)

    def _allowed_mask(self, paths):
        """[K, F] uint8 gate from interaction constraints (None if off).

        ``paths``: per node, the tuple of split features on its root
        path. A feature is allowed iff some constraint set contains the
        whole path and the feature (xgboost semantics; at the root every
        feature may open a path).
        """
        if self.interaction_sets is None:
            return None
        F = self.dtrain.n_features
        m = np.zeros((len(paths), F), np.uint8)
        for i, path in enumerate(paths):
            pf = set(path)
            if not pf:
                m[i, :] = 1  # root: any feature may open the path
            else:
                allow = set()
                for cs in self.interaction_sets:
                    if pf <= cs:
                        allow |= cs
                for f in allow:
                    if 0 <= f < F:
                        m[i, f] = 1
        return torch.from_numpy(m)

    def _grow_tree(
        self, gpair: torch.Tensor, it: int, cls: int, ptree: int = 0
    ) -> Tree:
        import os as _os
        import time as _time

        _prof = _os.environ.get("RXGB_PROFILE_ENGINE")
        _t = {}

        def _tick(name, _last=[None]):
            if not _prof:
                return
            if self.device.type == "cuda":
                torch.cuda.synchronize()
            now = _time.perf_counter()
            if _last[0] is not None:
                _t[name] = _t.get(name, 0.0) + now - _last[0]
            _last[0] = now

        _tick(None)
        gq, scale_g, scale_h = self._quantize(gpair)
        _tick("quantize")
        self._scale_g_cur = scale_g
        self._leaf_segs = []
        if self.p.grow_policy == "lossguide":
            return self._grow_tree_lossguide(gq, scale_g, scale_h, it, cls, ptree)
        ridx = self._sample_rows(it, cls + 101 * ptree)
        # segment-ordered gradient pairs aligned with ridx; the partition
        # scatter permutes them each depth, so no per-depth gather pass
        if ridx.numel() == self.dtrain.n_rows:
            gseg = gq  # identity order (partition clones before writing)
        else:
            gseg = gq[ridx.long()].contiguous()
        feat_mask = self._sample_features(it, cls + 101 * ptree)
        n_local = int(ridx.numel())

        if n_local == 0:
            root_sum = torch.zeros(2, dtype=torch.int64, device=self.device)
        elif n_local == self.dtrain.n_rows:
            # no row sampling: ridx is the identity - skip the per-row gather
            root_sum = gq.sum(dim=0, dtype=torch.int64)
        else:
            root_sum = gq[ridx.long()].sum(dim=0, dtype=torch.int64)
        if self.coll.is_distributed:
            self.coll.allreduce_(root_sum)

        # ---- SoA tree storage (grow-doubling numpy arrays; the previous
        # python-list-per-field representation plus _Node objects cost
        # ~8 ms/round of host bookkeeping at depth 12 / 8191 nodes)
        ta = _TreeArrays(max(64, 2 ** min(self.p.max_depth + 2, 16)))
        cuts_flat_cpu = self.dtrain.cuts.cuts_flat.cpu().numpy()
        cut_ptr_cpu = self.dtrain.cuts.cut_ptr.cpu().numpy()
        track_paths = self.interaction_sets is not None
        mono_np = (
            self.mono.cpu().numpy().astype(np.int64)
            if self.mono is not None else None
        )

        # ---- SoA frontier: one numpy array per field. Children are
        # always appended as (left, right) pairs, so siblings sit at
        # positions 2i / 2i+1 and the build/derive pairing vectorizes.
        fr_nid = np.zeros(1, np.int64)
        fr_start = np.zeros(1, np.int64)
        fr_count = np.array([n_local], np.int64)
        fr_sumg = np.array([int(root_sum[0])], np.int64)
        fr_sumh = np.array([int(root_sum[1])], np.int64)
        fr_pslot = np.zeros(1, np.int64)
        fr_wlo = np.full(1, -np.inf)
        fr_whi = np.full(1, np.inf)
        fr_paths = [()] if track_paths else None

        import os as _os2

        # single-sync fused loop decision is per-tree (see the in-loop
        # comment); the ping-pong ridx buffers exist ONLY for it - the
        # 2-sync path reassigns ridx to fresh clones each depth, which
        # would leave buffer 0 stale for the margin update
        use_fused_loop = (
            self.device.type == "cuda"
            and _os2.environ.get("RXGB_ONE_SYNC", "1") != "0"
        )
        ridx_bufs = None
        cur_buf = 0
        if use_fused_loop:
            ridx_bufs = [ridx, torch.empty_like(ridx)]

        lam = float(self.p.reg_lambda)
        alpha = float(self.p.reg_alpha)
        prev_all_hist: Optional[torch.Tensor] = None
        for depth in range(self.p.max_depth):
            nF = fr_nid.size
            if nF == 0:
                break
            # ---- build order: per sibling pair build the globally
            # smaller child (quantized hessian sum as the size proxy -
            # identical on every rank; local counts are not), derive the
            # sibling by subtraction. Built block = contiguous leading
            # slice (one AllReduce, batched subtraction for the rest).
            if depth == 0:
                order_idx = np.zeros(1, np.int64)
                K = 1
                derive_pslot = np.zeros(0, np.int64)
                derive_sib_pos = np.zeros(0, np.int64)
            else:
                ia = np.arange(0, nF, 2)
                ib = ia + 1
                small_is_a = fr_sumh[ia] <= fr_sumh[ib]
                idx_small = np.where(small_is_a, ia, ib)
                idx_big = np.where(small_is_a, ib, ia)
                order_idx = np.concatenate([idx_small, idx_big])
                K = idx_small.size
                derive_pslot = fr_pslot[idx_big]
                derive_sib_pos = np.arange(K, dtype=np.int64)

            # scan-slot order: built block first, derived siblings after
            nid_ord = fr_nid[order_idx]
            start_ord = fr_start[order_idx]
            count_ord = fr_count[order_idx]
            sumg_ord = fr_sumg[order_idx]
            sumh_ord = fr_sumh[order_idx]
            wlo_ord = fr_wlo[order_idx]
            whi_ord = fr_whi[order_idx]
            paths_ord = (
                [fr_paths[i] for i in order_idx] if track_paths else None
            )

            starts = torch.from_numpy(np.ascontiguousarray(start_ord[:K]))
            counts = torch.from_numpy(np.ascontiguousarray(count_ord[:K]))
            KK = sumg_ord.size
            nd = derive_sib_pos.size

            # feature gate needed by both the python path and depth_step
            fb = self.feat_bins
            mask = feat_mask
            if self.p.colsample_bylevel < 1.0:
                lvl_mask = self._sample_features(it, cls, level=depth)
                mask = lvl_mask if mask is None else (mask & lvl_mask)
            if mask is not None:
                fb = torch.where(
                    mask, self.feat_bins, torch.zeros_like(self.feat_bins)
                )
            allowed_m = (
                self._allowed_mask(paths_ord) if track_paths else None
            )
            if self.p.colsample_bynode < 1.0:
                # per-node feature sampling (xgboost colsample_bynode):
                # deterministic in (seed, iteration, class, depth) and
                # in scan-slot order, so every rank draws the same masks
                node_m = self._bynode_mask(KK, it, cls, depth)
                allowed_m = (
                    node_m if allowed_m is None
                    else (allowed_m.to(torch.uint8) & node_m)
                )

            F = self.dtrain.n_features
            all_hist = torch.empty(
                (nF, F, self.n_bins, 2), dtype=torch.int64, device=self.device
            )

            # depth_step (one C++ call per depth) measured slightly
            # SLOWER than the python-orchestrated fused loop in a
            # same-box interleaved A/B (3.23-3.36 vs 3.20-3.21
            # ms/round): per-depth python/torch dispatch is not on the
            # critical path - the residual wall gap is event-wait and
            # launch bubbles. Kept opt-in (RXGB_DEPTH_STEP=1).
            use_depth_step = (
                use_fused_loop and not self.coll.is_distributed
                and _os2.environ.get("RXGB_DEPTH_STEP") == "1"
            )
            if use_depth_step:
                # one C++ call per depth: stage + zero + hist + derive +
                # scan + device-planned partition + pull + scatter. The
                # ~30 per-depth python/torch dispatches were ~0.8 ms of
                # wall per round at depth 8.
                dm_t = torch.from_numpy(
                    np.concatenate(
                        [sumg_ord, sumh_ord, start_ord, count_ord,
                         derive_pslot, derive_sib_pos]
                    )
                )
                # C++ find_splits takes raw pointers: everything must
                # already be the right dtype ON DEVICE
                mono_t = (
                    self.mono.to(self.device).to(torch.int8)
                    if self.mono is not None
                    else torch.zeros(0, dtype=torch.int8,
                                     device=self.device)
                )
                bounds_t = (
                    torch.from_numpy(np.ascontiguousarray(
                        np.stack([wlo_ord, whi_ord], axis=1)
                    )).to(self.device)
                    if self.mono is not None
                    else torch.zeros((0, 2), dtype=torch.float64,
                                     device=self.device)
                )
                allowed_t = (
                    allowed_m.to(self.device).to(torch.uint8).contiguous()
                    if allowed_m is not None
                    else torch.zeros(0, dtype=torch.uint8,
                                     device=self.device)
                )
                bins_t_t = getattr(self.dtrain, "bins_t", None)
                if bins_t_t is None:
                    bins_t_t = torch.zeros(0, dtype=torch.uint8,
                                           device=self.device)
                gseg_in = (
                    gseg if gseg is not None
                    else torch.zeros((0, 2), dtype=torch.int32,
                                     device=self.device)
                )
                frontier_rows = int(count_ord.sum())
                chunk_bound = (frontier_rows + 2047) // 2048 + KK
                from xgboost_ray_amd.ops import gpu as _gpu

                ridx, gseg, pull = _gpu._load().depth_step(
                    self.dtrain.bins, bins_t_t, gseg_in, ridx,
                    ridx_bufs[1 - cur_buf],
                    prev_all_hist if prev_all_hist is not None
                    else all_hist,
                    all_hist, dm_t, KK, int(nd), K, starts, counts,
                    fb.to(self.device).to(torch.int32), scale_g,
                    scale_h, self.p.reg_lambda,
                    self.p.reg_alpha, self.p.gamma,
                    self.p.min_child_weight, mono_t, bounds_t,
                    allowed_t, self.n_bins, chunk_bound,
                )
                depth_in_buf = cur_buf
                cur_buf = 1 - cur_buf
                _tick("scan")
                _gpu._load().wait_pull_event()
                arr = pull[: 7 * KK].numpy()
                pk = arr[: 6 * KK].reshape(KK, 6)
                gain = pk[:, 0].astype(np.int32).view(np.float32).copy()
                bfeat = pk[:, 1].astype(np.int32)
                bbin = pk[:, 2].astype(np.int32)
                bdl = pk[:, 3].astype(np.uint8)
                blg = pk[:, 4].copy()
                blh = pk[:, 5].copy()
                lc_full = arr[6 * KK : 7 * KK].copy()
                use_fused = True
                _tick("scan_pull")
                return_to_decode = True
            else:
                return_to_decode = False

            if not return_to_decode:
                hist = all_hist[:K]  # built block (contiguous slice)
                hist.zero_()  # derived rows overwritten by the sub
                overlap = self.coll.is_distributed and (
                    (
                        self.device.type == "cuda"
                        and self.dtrain.bins.stride(0) % 16 == 0
                        and F > 16
                    )
                    or _os2.environ.get("RXGB_FORCE_OVERLAP_ALLREDUCE") == "1"
                )
                if overlap:
                    # overlap the RCCL AllReduce of each finished feature block
                    # with the build of the next (BASELINE north-star: side-
                    # stream collective over xGMI behind the histogram build)
                    n_chunks = min(4, (F + 31) // 32)
                    step = ((F + n_chunks - 1) // n_chunks + 15) // 16 * 16
                    pending = []
                    f0 = 0
                    while f0 < F:
                        f1 = min(f0 + step, F)
                        ops.build_histogram(
                            self.dtrain.bins, gseg, ridx, starts, counts,
                            self.n_bins, f_range=(f0, f1), out=hist,
                            pregathered=True,
                        )
                        sl = hist[:, f0:f1].contiguous()
                        pending.append((self.coll.allreduce_async(sl), sl, f0, f1))
                        f0 = f1
                    for h, sl, c0, c1 in pending:
                        h.wait()
                        hist[:, c0:c1].copy_(sl)
                else:
                    ops.build_histogram(
                        self.dtrain.bins, gseg, ridx, starts, counts,
                        self.n_bins, out=hist, pregathered=True,
                    )
                    _tick("hist")
                    if self.coll.is_distributed:
                        self.coll.allreduce_(hist)
                _tick("allreduce")
                # ONE H2D for this depth's control data (layout in comment
                # above depth_step)
                if self.device.type == "cuda":
                    depth_meta = self._stage_i64(
                        np.concatenate(
                            [sumg_ord, sumh_ord, start_ord, count_ord,
                             derive_pslot, derive_sib_pos]
                        ),
                        "depth_meta",
                    )
                    pg, ph = depth_meta[:KK], depth_meta[KK : 2 * KK]
                    d_starts = depth_meta[2 * KK : 3 * KK]
                    d_counts = depth_meta[3 * KK : 4 * KK]
                    pslots = depth_meta[4 * KK : 4 * KK + nd]
                    spos = depth_meta[4 * KK + nd :]
                else:
                    pg = torch.from_numpy(sumg_ord)
                    ph = torch.from_numpy(sumh_ord)
                    pslots = torch.from_numpy(derive_pslot)
                    spos = torch.from_numpy(derive_sib_pos)
                if nd:
                    # sibling = parent - built, batched over all pairs
                    torch.sub(
                        prev_all_hist.index_select(0, pslots),
                        all_hist.index_select(0, spos),
                        out=all_hist[K:],
                    )
                _tick("stack")
                mono_bounds = None
                if self.mono is not None:
                    mono_bounds = torch.from_numpy(
                        np.ascontiguousarray(
                            np.stack([wlo_ord, whi_ord], axis=1)
                        )
                    )
                # single-sync fused path (GPU, default): the split scan's
                # packed output feeds a device-planned partition directly;
                # ONE pinned D2H brings {splits | left_counts} to the host,
                # which replays the identical split predicate for
                # bookkeeping. The pull is enqueued BEFORE the scatter and
                # the host waits on an inter-launch EVENT, so it wakes
                # right after count+prefix and every piece of bookkeeping
                # overlaps the scatter: measured d8 3.56-3.61 vs 3.72-3.73
                # ms/round for the 2-sync structure (the first version
                # synced past the scatter and was SLOWER - see
                # profiles/README.md pass 5/8). RXGB_ONE_SYNC=0 restores
                # the 2-sync path.
                use_fused = use_fused_loop
                lc_full = None
                if use_fused:
                    packed_dev = ops.find_splits(
                        all_hist, pg, ph, fb, scale_g, scale_h,
                        self.p.reg_lambda, self.p.reg_alpha, self.p.gamma,
                        self.p.min_child_weight, monotone=self.mono,
                        bounds=mono_bounds, allowed=allowed_m, pull=False,
                    )
                    _tick("scan")
                    # grid BOUND: every split node adds at most one
                    # partial chunk (PART_CHUNK = 2048 rows/workgroup)
                    frontier_rows = int(count_ord.sum())
                    chunk_bound = (frontier_rows + 2047) // 2048 + KK
                    ridx, gseg, pull, pull_ev = (
                        ops.partition_rows_from_packed(
                            self.dtrain.bins, ridx, d_starts, d_counts,
                            packed_dev, gseg,
                            getattr(self.dtrain, "bins_t", None),
                            chunk_bound,
                            ridx_dest=ridx_bufs[1 - cur_buf],
                        )
                    )
                    depth_in_buf = cur_buf
                    cur_buf = 1 - cur_buf
                    # event recorded between the pull D2H and the scatter:
                    # the host wakes before the scatter and every piece of
                    # bookkeeping below overlaps it
                    pull_ev.synchronize()
                    arr = pull[: 7 * KK].numpy()
                    pk = arr[: 6 * KK].reshape(KK, 6)
                    gain = pk[:, 0].astype(np.int32).view(np.float32).copy()
                    bfeat = pk[:, 1].astype(np.int32)
                    bbin = pk[:, 2].astype(np.int32)
                    bdl = pk[:, 3].astype(np.uint8)
                    blg = pk[:, 4].copy()
                    blh = pk[:, 5].copy()
                    lc_full = arr[6 * KK : 7 * KK].copy()
                    _tick("scan_pull")
                else:
                    best = ops.find_splits(
                        all_hist,
                        pg,
                        ph,
                        fb,
                        scale_g,
                        scale_h,
                        self.p.reg_lambda,
                        self.p.reg_alpha,
                        self.p.gamma,
                        self.p.min_child_weight,
                        monotone=self.mono,
                        bounds=mono_bounds,
                        allowed=allowed_m,
                    )
                    _tick("scan")
                    gain = best["gain"]
                    _tick("scan_pull")
                    bfeat = best["feature"]
                    bbin = best["bin"]
                    bdl = best["default_left"]
                    blg = best["left_g"]
                    blh = best["left_h"]

            # ---- vectorized split/leaf decision for the whole frontier
            splits_ok = (gain > 0) & (bfeat >= 0) & np.isfinite(gain)
            okf = np.nonzero(splits_ok)[0]  # ascending == scan-slot order
            leaf_idx = np.nonzero(~splits_ok)[0]
            n_split = int(okf.size)

            if n_split == 0:
                if leaf_idx.size:
                    self._finalize_leaves_soa(
                        ta, nid_ord[leaf_idx], sumg_ord[leaf_idx],
                        sumh_ord[leaf_idx], start_ord[leaf_idx],
                        count_ord[leaf_idx], wlo_ord[leaf_idx],
                        whi_ord[leaf_idx], scale_h,
                        parity=(depth_in_buf if use_fused else 0),
                    )
                fr_nid = np.zeros(0, np.int64)
                break

            # launch the partition's count+prefix FIRST (GPU): the leaf
            # finalization and tree-array writes below then run on the
            # host while those kernels execute, and the children
            # bookkeeping after finish() overlaps the scatter
            sf32 = bfeat[okf].astype(np.int32)
            sb32 = bbin[okf].astype(np.int32)
            sdl8 = bdl[okf].astype(np.uint8)
            part_ctx = None
            if lc_full is None and self.device.type == "cuda":
                part_ctx = ops.partition_begin(
                    self.dtrain.bins,
                    ridx,
                    torch.from_numpy(np.ascontiguousarray(start_ord[okf])),
                    torch.from_numpy(np.ascontiguousarray(count_ord[okf])),
                    torch.from_numpy(sf32),
                    torch.from_numpy(sb32),
                    torch.from_numpy(sdl8),
                    gpair_seg=gseg,
                    bins_t=getattr(self.dtrain, "bins_t", None),
                )

            if leaf_idx.size:
                self._finalize_leaves_soa(
                    ta, nid_ord[leaf_idx], sumg_ord[leaf_idx],
                    sumh_ord[leaf_idx], start_ord[leaf_idx],
                    count_ord[leaf_idx], wlo_ord[leaf_idx],
                    whi_ord[leaf_idx], scale_h,
                    parity=(depth_in_buf if use_fused else 0),
                )

            # child ids in scan-slot order (matches the sequential
            # numbering the list-based builder produced)
            base = ta.n
            lids = base + 2 * np.arange(n_split, dtype=np.int64)
            ta.ensure(base + 2 * n_split)
            nids_ok = nid_ord[okf]
            f_ok = bfeat[okf].astype(np.int64)
            b_ok = bbin[okf].astype(np.int64)
            # split after bin b: left iff bin <= b; bin b covers
            # [cut[b-1], cut[b]) so "bin <= b" <=> v < cut[b] = thr.
            thr_ok = cuts_flat_cpu[cut_ptr_cpu[f_ok] + b_ok].astype(
                np.float64
            )
            ta.feat[nids_ok] = f_ok
            ta.thr[nids_ok] = thr_ok
            ta.left[nids_ok] = lids
            ta.dl[nids_ok] = bdl[okf]
            ta.gain[nids_ok] = gain[okf].astype(np.float64)
            ta.cover[nids_ok] = sumh_ord[okf].astype(np.float64) / scale_h
            ta.parent[lids] = nids_ok
            ta.parent[lids + 1] = nids_ok
            ta.n = base + 2 * n_split

            _tick("tree_host")
            if lc_full is not None:
                # fused path: the partition already ran on device; its
                # per-split-node left counts rode the single pull, in
                # the same ascending-okf order the plan kernel compacted
                lc = lc_full[:n_split].astype(np.int64)
            elif part_ctx is not None:
                ridx, left_counts, gseg = ops.partition_finish(part_ctx)
                lc = left_counts.numpy().astype(np.int64)
            else:
                sstarts = torch.from_numpy(
                    np.ascontiguousarray(start_ord[okf])
                )
                scounts = torch.from_numpy(
                    np.ascontiguousarray(count_ord[okf])
                )
                ridx, left_counts, gseg = ops.partition_rows(
                    self.dtrain.bins,
                    ridx,
                    sstarts,
                    scounts,
                    torch.from_numpy(sf32),
                    torch.from_numpy(sb32),
                    torch.from_numpy(sdl8),
                    gpair_seg=gseg,
                    bins_t=getattr(self.dtrain, "bins_t", None),
                )
                lc = left_counts.cpu().numpy().astype(np.int64)
            _tick("partition")
            l_start = start_ord[okf]
            l_sumg = blg[okf]
            l_sumh = blh[okf]
            r_start = l_start + lc
            r_count = count_ord[okf] - lc
            r_sumg = sumg_ord[okf] - l_sumg
            r_sumh = sumh_ord[okf] - l_sumh

            # monotone child bounds (exact op order of the scalar path)
            wlo_p = wlo_ord[okf]
            whi_p = whi_ord[okf]
            l_wlo, l_whi = wlo_p.copy(), whi_p.copy()
            r_wlo, r_whi = wlo_p.copy(), whi_p.copy()
            if mono_np is not None:
                cvec = mono_np[sf32]
                act = cvec != 0
                if act.any():
                    wl = np.maximum(wlo_p, np.minimum(
                        whi_p, _calc_weight_vec(
                            l_sumg / scale_g, l_sumh / scale_h, lam, alpha
                        )))
                    wr = np.maximum(wlo_p, np.minimum(
                        whi_p, _calc_weight_vec(
                            r_sumg / scale_g, r_sumh / scale_h, lam, alpha
                        )))
                    mid = 0.5 * (wl + wr)
                    pos = act & (cvec > 0)
                    neg = act & (cvec < 0)
                    l_whi = np.where(pos, np.minimum(l_whi, mid), l_whi)
                    r_wlo = np.where(pos, np.maximum(r_wlo, mid), r_wlo)
                    l_wlo = np.where(neg, np.maximum(l_wlo, mid), l_wlo)
                    r_whi = np.where(neg, np.minimum(r_whi, mid), r_whi)

            n2 = 2 * n_split
            fr_nid = np.empty(n2, np.int64)
            fr_nid[0::2] = lids
            fr_nid[1::2] = lids + 1
            fr_start = np.empty(n2, np.int64)
            fr_start[0::2] = l_start
            fr_start[1::2] = r_start
            fr_count = np.empty(n2, np.int64)
            fr_count[0::2] = lc
            fr_count[1::2] = r_count
            fr_sumg = np.empty(n2, np.int64)
            fr_sumg[0::2] = l_sumg
            fr_sumg[1::2] = r_sumg
            fr_sumh = np.empty(n2, np.int64)
            fr_sumh[0::2] = l_sumh
            fr_sumh[1::2] = r_sumh
            fr_pslot = np.empty(n2, np.int64)
            fr_pslot[0::2] = okf
            fr_pslot[1::2] = okf
            fr_wlo = np.empty(n2, np.float64)
            fr_wlo[0::2] = l_wlo
            fr_wlo[1::2] = r_wlo
            fr_whi = np.empty(n2, np.float64)
            fr_whi[0::2] = l_whi
            fr_whi[1::2] = r_whi
            if track_paths:
                new_paths = []
                for k_i, f_i in zip(okf.tolist(), sf32.tolist()):
                    cp = tuple(sorted(set(paths_ord[k_i]) | {int(f_i)}))
                    new_paths.append(cp)
                    new_paths.append(cp)
                fr_paths = new_paths
            prev_all_hist = all_hist

        # remaining frontier nodes (max depth reached) become leaves
        if fr_nid.size:
            self._finalize_leaves_soa(
                ta, fr_nid, fr_sumg, fr_sumh, fr_start, fr_count,
                fr_wlo, fr_whi, scale_h, parity=cur_buf,
            )
        # leaf margin update: every row's final node is its segment's
        # node; _finalize_leaves_soa collected (start, count, value) segs
        leaf_starts = [s for (s, c, v, bp) in self._leaf_segs]
        leaf_counts = [c for (s, c, v, bp) in self._leaf_segs]
        leaf_vals = [v for (s, c, v, bp) in self._leaf_segs]
        leaf_parity = [bp for (s, c, v, bp) in self._leaf_segs]
        _tick("leaf_host")

        n = ta.n
        tree = Tree(
            feat=ta.feat[:n].astype(np.int32),
            thr=ta.thr[:n].astype(np.float32),
            left=ta.left[:n].astype(np.int32),
            default_left=ta.dl[:n].astype(np.uint8),
            value=ta.val[:n].astype(np.float32),
            gain=ta.gain[:n].astype(np.float32),
            cover=ta.cover[:n].astype(np.float32),
            parent=np.where(
                ta.parent[:n] < 0, 2147483647, ta.parent[:n]
            ).astype(np.int32),
        )

        if n_local < self.dtrain.n_rows:
            # row sampling: EVERY row's margin must advance (xgboost
            # UpdatePredictionCache semantics) - the leaf-segment
            # scatter only covers the sampled rows, which left
            # out-of-sample gradients on stale margins (fuzz-found in
            # round 2; also broke resume==continuous under subsample).
            # The binned tree walk lands each row in the identical leaf
            # the partition chose, so sampled rows get the same bits.
            mv = self.margin if self.n_class == 1 else self.margin[:, cls]
            self._add_tree_margin_raw(tree, mv)
        elif leaf_starts:
            ops.update_margins(
                self.margin if self.n_class == 1 else self.margin[:, cls],
                ridx_bufs[0] if ridx_bufs is not None else ridx,
                torch.tensor(leaf_starts, dtype=torch.int64),
                torch.tensor(leaf_counts, dtype=torch.int64),
                np.asarray(leaf_vals, dtype=np.float32),
                ridx_b=(ridx_bufs[1] if ridx_bufs is not None else None),
                parity=(leaf_parity if ridx_bufs is not None else None),
            )
        self._leaf_segs = []
        _tick("margins")
        if _prof:
            print("PROF", {k: round(v*1000, 2) for k, v in _t.items()}, flush=True)
        return tree

    _leaf_segs: List[Tuple[int, int, float]] = []


    def _finalize_leaves_soa(self, ta, nids, sumg, sumh, starts, counts,
                             wlo, whi, scale_h, parity=0):
        """SoA leaf finalization: same math/op order as
        _finalize_leaves_batch, writing into _TreeArrays and collecting
        (start, count, value) margin segments."""
        if nids.size == 0:
            return
        G = sumg.astype(np.float64) / self._scale_g_cur
        H = sumh.astype(np.float64) / scale_h
        lam = self.p.reg_lambda
        alpha = self.p.reg_alpha
        denom = H + lam
        Gs = G
        if alpha > 0:
            Gs = np.copysign(np.maximum(np.abs(G) - alpha, 0.0), G)
        w = np.where(denom > 0, -Gs / np.where(denom > 0, denom, 1.0), 0.0)
        if self.mono is not None:
            w = np.clip(w, wlo, whi)
        if self.p.max_delta_step > 0:
            w = np.clip(w, -self.p.max_delta_step, self.p.max_delta_step)
        v = self.p.eta * w
        ta.val[nids] = v
        ta.cover[nids] = H
        self._leaf_segs.extend(
            zip(starts.tolist(), counts.tolist(), v.tolist(),
                [parity] * int(nids.size))
        )

    def _finalize_leaves_batch(self, nodes, val_l, cover_l, scale_h):
        """Vectorized _finalize_leaf for a whole frontier (the per-node
        Python loop was 5.6 ms/round on 4096-leaf depth-12 trees).
        Same op order as _calc_weight, so values are bit-identical."""
        if not nodes:
            return
        G = np.fromiter(
            (nd.sum_g for nd in nodes), np.float64, len(nodes)
        ) / self._scale_g_cur
        H = np.fromiter(
            (nd.sum_h for nd in nodes), np.float64, len(nodes)
        ) / scale_h
        lam = self.p.reg_lambda
        alpha = self.p.reg_alpha
        denom = H + lam
        Gs = G
        if alpha > 0:
            Gs = np.copysign(np.maximum(np.abs(G) - alpha, 0.0), G)
        w = np.where(denom > 0, -Gs / np.where(denom > 0, denom, 1.0), 0.0)
        if self.mono is not None:
            w = np.clip(
                w,
                np.fromiter((nd.w_lower for nd in nodes), np.float64,
                            len(nodes)),
                np.fromiter((nd.w_upper for nd in nodes), np.float64,
                            len(nodes)),
            )
        if self.p.max_delta_step > 0:
            w = np.clip(w, -self.p.max_delta_step, self.p.max_delta_step)
        v = self.p.eta * w
        segs = self._leaf_segs
        for nd, vi, Hi in zip(nodes, v.tolist(), H.tolist()):
            val_l[nd.nid] = vi
            cover_l[nd.nid] = Hi
            segs.append((nd.start, nd.count, vi, 0))

    def _finalize_leaf(self, nd: _Node, val_l, cover_l, scale_h):
        G = nd.sum_g / self._scale_g_cur
        H = nd.sum_h / scale_h
        w = _calc_weight(G, H, self.p.reg_lambda, self.p.reg_alpha)
        if self.mono is not None:
            w = _clamp(w, nd.w_lower, nd.w_upper)
        if self.p.max_delta_step > 0:
            w = max(-self.p.max_delta_step, min(self.p.max_delta_step, w))
        val_l[nd.nid] = self.p.eta * w
        cover_l[nd.nid] = H
        self._leaf_segs.append((nd.start, nd.count, self.p.eta * w, 0))

    # -- evaluation ----------------------------------------------------------
    def eval_sets(self, evals: Sequence[EvalPack], feval=None) -> Dict[str, Dict[str, float]]:
        metric_names = list(self.p.eval_metric)
        if not metric_names and not self.p.disable_default_eval_metric:
            metric_names = [self.obj.default_metric]
        out: Dict[str, Dict[str, float]] = {}
        local_stats = []
        specs = []
        for ev in evals:
            margin = self.margin if ev.X is None else ev.margin
            label = self._train_label if ev.X is None else ev.label
            weight = self.dtrain.weight if ev.X is None else ev.weight
            qid = self.dtrain.qid if ev.X is None else ev.qid
            for mname in metric_names:
                m = get_metric(mname)
                st = m.local_stats(margin, label, weight, qid, self.obj)
                local_stats.append(st)
                specs.append((ev.name, m))
        if specs:
            flat = torch.cat([s.flatten() for s in local_stats])
            if self.coll.is_distributed:
                flat_d = flat.to(self.device) if self.device.type == "cuda" else flat
                self.coll.allreduce_(flat_d)
                flat = flat_d.cpu()
            pos = 0
            for (ename, m), st in zip(specs, local_stats):
                sz = st.numel()
                val = m.finalize(flat[pos : pos + sz])
                pos += sz
                out.setdefault(ename, {})[m.name] = val
        if feval is not None:
            for ev in evals:
                margin = self.margin if ev.X is None else ev.margin
                label = self.dtrain.label if ev.X is None else ev.label

                class _Shim:
                    def __init__(self, l):
                        self._l = l

                    def get_label(self):
                        return self._l

                name, val = feval(
                    margin.cpu().numpy(),
                    _Shim(None if label is None else label.cpu().numpy()),
                )
                val = float(val)
                if self.coll.is_distributed:
                    # a custom feval sees only the local shard; average
                    # across ranks so every rank's early-stopping decision
                    # is identical (rank-divergent stops break collectives)
                    t = torch.tensor(
                        [val], dtype=torch.float64,
                        device=self.device
                        if self.device.type == "cuda" else "cpu",
                    )
                    self.coll.allreduce_(t)
                    val = float(t[0]) / self.coll.world_size
                out.setdefault(ev.name, {})[name] = val
        return out

    def update_eval_margins(self, evals: Sequence[EvalPack], trees, classes):
        """Incrementally add the new round's trees to eval-set margins."""
        if self.p.booster == "dart":
            # _dart_commit rescales previously-added and dropped trees'
            # leaf values in place every round, so incremental margins
            # would drift from the real model (eval logloss reported
            # ~0.18 where the true model scored ~0.24 in a 20-round run).
            # Recompute from the booster: exact by construction, and dart
            # eval sets are small relative to training data.
            for ev in evals:
                if ev.X is None:
                    continue
                pm = self.booster.predict_margin_tensor(ev.X).to(
                    torch.float32
                )
                if ev.base_margin is not None:
                    pm = pm + ev.base_margin.reshape(pm.shape)
                ev.margin = pm
            return
        if self.p.booster == "gblinear":
            for ev in evals:
                if ev.X is None:
                    continue
                for cls, (dwf, db) in enumerate(self._lin_deltas):
                    mv = (ev.margin if self.n_class == 1
                          else ev.margin[:, cls])
                    mv += ev.X @ dwf.to(ev.X.device)
                    mv += db
            return
        for ev in evals:
            if ev.X is None:
                continue
            for t, cls in zip(trees, classes):
                mv = ev.margin if self.n_class == 1 else ev.margin[:, cls]
                flat_feat = torch.from_numpy(t.feat).to(ev.X.device)
                ops.predict_trees(
                    ev.X,
                    flat_feat,
                    torch.from_numpy(t.thr).to(ev.X.device),
                    torch.from_numpy(t.left).to(ev.X.device),
                    torch.from_numpy(t.default_left).to(ev.X.device),
                    torch.from_numpy(t.value).to(ev.X.device),
                    torch.tensor([0, t.num_nodes], dtype=torch.int32),
                    mv,
                )

    def init_eval_margins(self, evals: Sequence[EvalPack], model: Optional[Booster]):
        for ev in evals:
            if ev.X is None:
                continue
            n = ev.X.shape[0]
            shape = (n,) if self.n_class == 1 else (n, self.n_class)
            ev.margin = torch.full(
                shape,
                self.base_margin_const,
                dtype=torch.float32,
                device=ev.X.device,
            )
            if ev.base_margin is not None:
                ev.margin += ev.base_margin.reshape(shape)
            if model is not None and model.trees:
                pm = model.predict_margin_tensor(ev.X)
                ev.margin = pm.to(torch.float32) + (
                    ev.base_margin.reshape(shape) if ev.base_margin is not None else 0
                )

    def _grow_tree_lossguide(self, gq, scale_g, scale_h, it, cls, ptree):
        """Leaf-wise (best-gain-first) growth - grow_policy=lossguide with
        max_leaves (XGBoost/LightGBM semantics). Each expansion partitions
        one leaf segment and histograms the smaller child (sibling by
        subtraction); the expansion order is driven by the globally
        identical scan gains, so all ranks grow the same tree."""
        import heapq

        ridx = self._sample_rows(it, cls + 101 * ptree)
        if ridx.numel() == self.dtrain.n_rows:
            gseg = gq
        else:
            gseg = gq[ridx.long()].contiguous()
        n_local = int(ridx.numel())
        if n_local == 0:
            root_sum = torch.zeros(2, dtype=torch.int64, device=self.device)
        elif n_local == self.dtrain.n_rows:
            root_sum = gq.sum(dim=0, dtype=torch.int64)
        else:
            root_sum = gq[ridx.long()].sum(dim=0, dtype=torch.int64)
        if self.coll.is_distributed:
            self.coll.allreduce_(root_sum)

        feat_mask = self._sample_features(it, cls + 101 * ptree)
        fb = self.feat_bins
        if feat_mask is not None:
            fb = torch.where(
                feat_mask, self.feat_bins, torch.zeros_like(self.feat_bins)
            )
        cuts_flat_cpu = self.dtrain.cuts.cuts_flat.cpu().numpy()
        cut_ptr_cpu = self.dtrain.cuts.cut_ptr.cpu().numpy()
        max_leaves = self.p.max_leaves or (1 << 30)
        max_depth = self.p.max_depth if self.p.max_depth > 0 else 64

        feat_l, thr_l = [-1], [0.0]
        left_l, dl_l, val_l, gain_l, cover_l = [-1], [0], [0.0], [0.0], [0.0]
        parent_l = [-1]

        def scan_nodes(hists, sums, paths=None):
            best = ops.find_splits(
                hists,
                torch.tensor([sg for sg, sh in sums], dtype=torch.int64,
                             device=self.device),
                torch.tensor([sh for sg, sh in sums], dtype=torch.int64,
                             device=self.device),
                fb, scale_g, scale_h, self.p.reg_lambda, self.p.reg_alpha,
                self.p.gamma, self.p.min_child_weight,
                allowed=self._allowed_mask(paths)
                if paths is not None and self.interaction_sets is not None
                else None,
            )
            return best

        # root
        root_hist = ops.build_histogram(
            self.dtrain.bins, gseg, ridx,
            torch.tensor([0], dtype=torch.int64),
            torch.tensor([n_local], dtype=torch.int64),
            self.n_bins, pregathered=True,
        )
        if self.coll.is_distributed:
            self.coll.allreduce_(root_hist)
        rbest = scan_nodes(root_hist, [(int(root_sum[0]), int(root_sum[1]))],
                           paths=[()])
        heap = []
        counter = [0]

        def push(nid, depth, start, count, sg, sh, hist, best_k, path=()):
            gain = float(best_k["gain"])
            rec = {
                "nid": nid, "depth": depth, "start": start, "count": count,
                "sg": sg, "sh": sh, "hist": hist, "path": path,
                "feature": int(best_k["feature"]), "bin": int(best_k["bin"]),
                "dl": int(best_k["default_left"]),
                "lg": int(best_k["left_g"]), "lh": int(best_k["left_h"]),
            }
            if gain > 0 and np.isfinite(gain) and rec["feature"] >= 0:
                heapq.heappush(heap, (-gain, counter[0], rec))
                counter[0] += 1
            else:
                self._finalize_leaf_rec(rec, val_l, cover_l, scale_h)

        def best_row(best, k):
            return {key: best[key][k] for key in best}

        push(0, 0, 0, n_local, int(root_sum[0]), int(root_sum[1]),
             root_hist[0], best_row(rbest, 0))
        n_leaves = 1
        while heap and n_leaves < max_leaves:
            neg_gain, _cnt, rec = heapq.heappop(heap)
            nid = rec["nid"]
            f, b = rec["feature"], rec["bin"]
            thr = float(cuts_flat_cpu[cut_ptr_cpu[f] + b])
            lid = len(feat_l)
            feat_l[nid] = f
            thr_l[nid] = thr
            left_l[nid] = lid
            dl_l[nid] = rec["dl"]
            gain_l[nid] = -neg_gain
            cover_l[nid] = rec["sh"] / scale_h
            for _i in range(2):
                feat_l.append(-1)
                thr_l.append(0.0)
                left_l.append(-1)
                dl_l.append(0)
                val_l.append(0.0)
                gain_l.append(0.0)
                cover_l.append(0.0)
                parent_l.append(nid)
            ridx, left_counts, gseg = ops.partition_rows(
                self.dtrain.bins, ridx,
                torch.tensor([rec["start"]], dtype=torch.int64),
                torch.tensor([rec["count"]], dtype=torch.int64),
                torch.tensor([f], dtype=torch.int32),
                torch.tensor([b], dtype=torch.int32),
                torch.tensor([rec["dl"]], dtype=torch.uint8),
                gpair_seg=gseg,
                bins_t=getattr(self.dtrain, "bins_t", None),
            )
            lcount = int(left_counts[0])
            lsums = (rec["lg"], rec["lh"])
            rsums = (rec["sg"] - rec["lg"], rec["sh"] - rec["lh"])
            lseg = (rec["start"], lcount)
            rseg = (rec["start"] + lcount, rec["count"] - lcount)
            children = [
                (lid, rec["depth"] + 1) + lseg + lsums,
                (lid + 1, rec["depth"] + 1) + rseg + rsums,
            ]
            # build the (globally) smaller child, derive the sibling
            small_i = 0 if lsums[1] <= rsums[1] else 1
            sc = children[small_i]
            sh_hist = ops.build_histogram(
                self.dtrain.bins, gseg, ridx,
                torch.tensor([sc[2]], dtype=torch.int64),
                torch.tensor([sc[3]], dtype=torch.int64),
                self.n_bins, pregathered=True,
            )
            if self.coll.is_distributed:
                self.coll.allreduce_(sh_hist)
            other_hist = rec["hist"] - sh_hist[0]
            rec["hist"] = None
            if small_i == 0:
                hists = torch.stack([sh_hist[0], other_hist])
            else:
                hists = torch.stack([other_hist, sh_hist[0]])
            child_path = tuple(sorted(set(rec.get("path", ())) | {f}))
            cbest = scan_nodes(
                hists, [(children[0][4], children[0][5]),
                        (children[1][4], children[1][5])],
                paths=[child_path, child_path],
            )
            for ci, ch in enumerate(children):
                if ch[1] >= max_depth:
                    self._finalize_leaf_rec(
                        {"start": ch[2], "count": ch[3], "sg": ch[4],
                         "sh": ch[5]}, val_l, cover_l, scale_h, nid=ch[0],
                    )
                else:
                    push(ch[0], ch[1], ch[2], ch[3], ch[4], ch[5],
                         hists[ci], best_row(cbest, ci), path=child_path)
            n_leaves += 1

        # drain remaining candidates as leaves
        while heap:
            _, _, rec = heapq.heappop(heap)
            self._finalize_leaf_rec(rec, val_l, cover_l, scale_h)

        leaf_starts = [st for (st, c, v, bp) in self._leaf_segs]
        leaf_counts = [c for (st, c, v, bp) in self._leaf_segs]
        leaf_vals = [v for (st, c, v, bp) in self._leaf_segs]
        tree = Tree(
            feat=np.asarray(feat_l, np.int32),
            thr=np.asarray(thr_l, np.float32),
            left=np.asarray(left_l, np.int32),
            default_left=np.asarray(dl_l, np.uint8),
            value=np.asarray(val_l, np.float32),
            gain=np.asarray(gain_l, np.float32),
            cover=np.asarray(cover_l, np.float32),
            parent=np.asarray(
                [2147483647 if pp < 0 else pp for pp in parent_l], np.int32
            ),
        )
        if int(ridx.numel()) < self.dtrain.n_rows:
            # row sampling: full-row margin advance (see depthwise note)
            mv = self.margin if self.n_class == 1 else self.margin[:, cls]
            self._add_tree_margin_raw(tree, mv)
        elif leaf_starts:
            ops.update_margins(
                self.margin if self.n_class == 1 else self.margin[:, cls],
                ridx,
                torch.tensor(leaf_starts, dtype=torch.int64),
                torch.tensor(leaf_counts, dtype=torch.int64),
                np.asarray(leaf_vals, dtype=np.float32),
            )
        self._leaf_segs = []
        return tree

    def _finalize_leaf_rec(self, rec, val_l, cover_l, scale_h, nid=None):
        nid = rec["nid"] if nid is None else nid
        G = rec["sg"] / self._scale_g_cur
        H = rec["sh"] / scale_h
        w = _calc_weight(G, H, self.p.reg_lambda, self.p.reg_alpha)
        if self.p.max_delta_step > 0:
            w = max(-self.p.max_delta_step, min(self.p.max_delta_step, w))
        val_l[nid] = self.p.eta * w
        cover_l[nid] = H
        self._leaf_segs.append(
            (rec["start"], rec["count"], self.p.eta * w, 0)
        )

    # quantization scale of the current tree (set in _grow_tree via _quantize)
    _scale_g_cur: float = 1.0


def _clamp(v: float, lo: float, up: float) -> float:
    return max(lo, min(up, v))


def _calc_weight(G: float, H: float, reg_lambda: float, reg_alpha: float) -> float:
    if H + reg_lambda <= 0:
        return 0.0
    if reg_alpha > 0:
        G = math.copysign(max(abs(G) - reg_alpha, 0.0), G)
    return -G / (H + reg_lambda)


def run_training(
    params: Dict,
    dtrain: BinnedMatrix,
    num_boost_round: int,
    evals: Sequence[EvalPack] = (),
    collective: Optional[Collective] = None,
    rank: int = 0,
    xgb_model: Optional[Booster] = None,
    callbacks=None,
    early_stopping_rounds: Optional[int] = None,
    verbose_eval: Union[bool, int] = False,
    obj: Optional[Callable] = None,
    feval: Optional[Callable] = None,
    maximize: Optional[bool] = None,
    evals_result: Optional[Dict] = None,
) -> Booster:
    """The per-rank training loop (reference: xgb.train inside
    RayXGBoostActor.train, reference main.py:722-752)."""
    engine = BoostingEngine(
        params, dtrain, collective, rank, custom_objective=obj
    )
    if xgb_model is not None:
        engine.load_model(xgb_model)
    engine.init_eval_margins(evals, xgb_model)
    cb = CallbackList(callbacks)
    log: Dict[str, Dict[str, List[float]]] = {}
    best_score, best_iter, stall = None, None, 0
    # explicit maximize (xgboost.train kwarg) wins; otherwise inferred
    # from the last builtin metric, defaulting to False for custom feval
    start_iter = engine.iteration

    for rnd in range(num_boost_round):
        it = engine.iteration
        if cb.before_iteration(engine.booster, it, log):
            break
        trees = engine.update()
        classes = engine.booster.tree_info[-len(trees):] if trees else []
        engine.update_eval_margins(evals, trees, classes)
        if evals:
            results = engine.eval_sets(evals, feval)
            for ename, md in results.items():
                for mname, v in md.items():
                    log.setdefault(ename, {}).setdefault(mname, []).append(v)
            if verbose_eval and rank == 0:
                interval = 1 if verbose_eval is True else int(verbose_eval)
                if it % interval == 0:
                    msg = "\t".join(
                        f"{e}-{m}:{v:.5f}"
                        for e, md in results.items()
                        for m, v in md.items()
                    )
                    print(f"[{it}]\t{msg}", flush=True)
            if early_stopping_rounds:
                last_eval = list(results.keys())[-1]
                last_metric = list(results[last_eval].keys())[-1]
                score = results[last_eval][last_metric]
                if maximize is None:
                    try:
                        maximize = get_metric(
                            last_metric.split("@")[0]
                        ).higher_better
                    except ValueError:
                        maximize = False  # custom feval: assume loss-like
                improved = (
                    best_score is None
                    or (score > best_score if maximize else score < best_score)
                )
                if improved:
                    best_score, best_iter, stall = score, it, 0
                else:
                    stall += 1
                    if stall >= early_stopping_rounds:
                        engine.booster.best_iteration = best_iter
                        engine.booster.best_score = best_score
                        engine.booster.set_attr(
                            best_iteration=str(best_iter),
                            best_score=str(best_score),
                        )
                        break
        if cb.after_iteration(engine.booster, it, log):
            break

    if best_iter is not None and engine.booster.best_iteration is None:
        engine.booster.best_iteration = best_iter
        engine.booster.best_score = best_score
        engine.booster.set_attr(
            best_iteration=str(best_iter), best_score=str(best_score)
        )
    if evals_result is not None:
        evals_result.update(log)
    cb.after_training(engine.booster, engine.iteration - 1, log)
    return engine.booster
