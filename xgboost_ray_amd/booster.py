"""Booster: the trained gradient-boosted-tree model.

Replaces the external ``xgboost.Booster`` the reference delegates to
(reference xgb.py:1-11, main.py:745). Stores trees as structure-of-arrays,
predicts through the device-dispatched tree-walk op, pickles cleanly for
driver-held checkpoints (reference main.py:612-626), and saves/loads the
XGBoost JSON model schema so models interoperate with stock XGBoost
(reference README.md:78 ``bst.save_model("model.xgb")``).
"""

import json
from dataclasses import dataclass
from typing import Dict, List, Optional, Sequence, Union

import numpy as np
import torch

from xgboost_ray_amd import ubjson


@dataclass
class Tree:
    """One regression tree, SoA layout.

    Node 0 is the root; children are allocated in pairs so
    ``right = left + 1``. ``feat < 0`` marks a leaf.
    """

    feat: np.ndarray  # int32 [n_nodes], -1 for leaves
    thr: np.ndarray  # float32 split threshold (go left iff x < thr)
    left: np.ndarray  # int32 left child (right = left + 1); -1 for leaves
    default_left: np.ndarray  # uint8
    value: np.ndarray  # float32 leaf weight (internal: base weight)
    gain: np.ndarray  # float32 loss_change for internal nodes
    cover: np.ndarray  # float32 sum_hessian
    parent: np.ndarray = None  # int32, -1 for root

    @property
    def num_nodes(self) -> int:
        return len(self.feat)


class DMatrix:
    """Local (single-process) data holder fed to the engine / predictor.

    The distributed equivalent is :class:`xgboost_ray_amd.matrix.RayDMatrix`;
    actors convert their shard into one of these
    (reference _get_dmatrix, main.py:379-445).
    """

    def __init__(
        self,
        data,
        label=None,
        weight=None,
        base_margin=None,
        qid=None,
        group=None,
        feature_names=None,
        missing=None,
    ):
        self.data = _as_float32_matrix(data, missing)
        self.label = _as_float32_vec(label)
        self.weight = _as_float32_vec(weight)
        self.base_margin = _as_float32_vec(base_margin)
        self.qid = None if qid is None else np.asarray(qid)
        if group is not None and qid is None:
            # convert group sizes to qid
            self.qid = np.repeat(np.arange(len(group)), np.asarray(group))
        self.feature_names = feature_names
        if feature_names is None and hasattr(data, "columns"):
            self.feature_names = [str(c) for c in data.columns]

    def num_row(self):
        return self.data.shape[0]

    def num_col(self):
        return self.data.shape[1]

    def get_label(self):
        return self.label


def _as_float32_matrix(data, missing=None):
    if data is None:
        return None
    if isinstance(data, torch.Tensor):
        data = data.detach().cpu().numpy()
    elif hasattr(data, "values") and not callable(data.values):  # pandas
        data = data.values
    arr = np.ascontiguousarray(np.asarray(data), dtype=np.float32)
    if arr.ndim == 1:
        arr = arr.reshape(-1, 1)
    if missing is not None and not (isinstance(missing, float) and np.isnan(missing)):
        arr = arr.copy()
        arr[arr == missing] = np.nan
    return arr


def _as_float32_vec(v):
    if v is None:
        return None
    if isinstance(v, torch.Tensor):
        v = v.detach().cpu().numpy()
    elif hasattr(v, "values") and not callable(v.values):
        v = v.values
    return np.ascontiguousarray(np.asarray(v), dtype=np.float32).reshape(-1)



def _fill_node_means(t, mean_val, cover):
    """Cover-weighted expected leaf value per node (iterative postorder)."""
    order = []
    stack = [0]
    while stack:
        nid = stack.pop()
        order.append(nid)
        if t.feat[nid] >= 0:
            stack.append(int(t.left[nid]))
            stack.append(int(t.left[nid]) + 1)
    for nid in reversed(order):
        f = t.feat[nid]
        if f < 0:
            mean_val[nid] = t.value[nid]
        else:
            l, r = int(t.left[nid]), int(t.left[nid]) + 1
            c = cover[nid] if cover[nid] > 0 else 1.0
            mean_val[nid] = (
                cover[l] * mean_val[l] + cover[r] * mean_val[r]
            ) / c


def _tree_shap(t, x, mean_val, cover, out, scale, condition=0,
               condition_feature=-1):
    """One row's SHAP contributions for one tree (Lundberg EXTEND/UNWIND).

    ``m`` is the path of unique features: fraction of zero paths (z),
    fraction of one paths (o), and feature index (d). ``condition`` +1
    fixes ``condition_feature`` present (its splits always follow x),
    -1 fixes it absent (cover-weighted mixture over its branches; the
    mixture weight travels as ``cf``) - the two passes that build SHAP
    interaction values. Conditioned splits never extend the path, so the
    conditioned feature receives no credit.
    """

    def extend(m, pz, po, pi):
        m = m + [[pz, po, pi, 1.0 if len(m) == 0 else 0.0]]
        ln = len(m)
        for i in range(ln - 2, -1, -1):
            m[i + 1][3] += po * m[i][3] * (i + 1) / ln
            m[i][3] = pz * m[i][3] * (ln - 1 - i) / ln
        return m

    def unwind_sum(m, i):
        ln = len(m) - 1
        po, pz = m[i][1], m[i][0]
        total = 0.0
        if po != 0.0:
            nxt = m[ln][3]
            for j in range(ln - 1, -1, -1):
                tmp = nxt * (ln + 1) / ((j + 1) * po)
                total += tmp
                nxt = m[j][3] - tmp * pz * (ln - j) / (ln + 1)
        else:
            for j in range(ln):
                total += m[j][3] * (ln + 1) / (pz * (ln - j))
        return total

    SKIP = -2  # sentinel: enter child without extending the path

    def recurse(nid, m, pz, po, pi, cf):
        if cf == 0.0:
            return
        if pi != SKIP:
            m = extend(list(map(list, m)), pz, po, pi)
        f = t.feat[nid]
        if f < 0:
            v = t.value[nid] * scale * cf
            for i in range(1, len(m)):
                w = unwind_sum(m, i)
                out[m[i][2]] += w * (m[i][1] - m[i][0]) * v
            return
        l, r = int(t.left[nid]), int(t.left[nid] + 1)
        fv = x[f]
        if np.isnan(fv):
            hot = l if t.default_left[nid] else r
        else:
            hot = l if fv < t.thr[nid] else r
        cold = r if hot == l else l
        c = cover[nid] if cover[nid] > 0 else 1.0
        if condition != 0 and f == condition_feature:
            if condition > 0:
                recurse(hot, m, 0.0, 0.0, SKIP, cf)
            else:
                recurse(hot, m, 0.0, 0.0, SKIP, cf * cover[hot] / c)
                recurse(cold, m, 0.0, 0.0, SKIP, cf * cover[cold] / c)
            return
        iz, io = 1.0, 1.0
        path_idx = next(
            (i for i in range(1, len(m)) if m[i][2] == f), None
        )
        if path_idx is not None:
            m = list(map(list, m))
            iz, io = m[path_idx][0], m[path_idx][1]
            ln = len(m) - 1
            po_, pz_ = m[path_idx][1], m[path_idx][0]
            nxt = m[ln][3]
            for j in range(ln - 1, -1, -1):
                if po_ != 0.0:
                    tmp = nxt * (ln + 1) / ((j + 1) * po_)
                    nxt = m[j][3] - tmp * pz_ * (ln - j) / (ln + 1)
                    m[j][3] = tmp
                else:
                    m[j][3] = m[j][3] * (ln + 1) / (pz_ * (ln - j))
            for j in range(path_idx, ln):
                m[j][0], m[j][1], m[j][2] = m[j + 1][0], m[j + 1][1], m[j + 1][2]
            m = m[:-1]
        recurse(hot, m, iz * cover[hot] / c, io, f, cf)
        recurse(cold, m, iz * cover[cold] / c, 0.0, f, cf)

    if condition == 0:
        out[len(x)] += mean_val[0] * scale
    recurse(0, [], 1.0, 1.0, -1, 1.0)


class Booster:
    def __init__(
        self,
        params: Optional[Dict] = None,
        trees: Optional[List[Tree]] = None,
        tree_info: Optional[List[int]] = None,
    ):
        self.params = dict(params or {})
        self.trees: List[Tree] = trees or []
        # class id each tree contributes to (all 0 for non-multiclass)
        self.tree_info: List[int] = tree_info or [0] * len(self.trees)
        self.attributes_: Dict[str, str] = {}
        self.best_iteration: Optional[int] = None
        self.best_score: Optional[float] = None
        self.feature_names: Optional[List[str]] = self.params.pop(
            "feature_names", None
        )
        # gblinear state: [F+1, num_group] float64, bias row last
        self.linear_weights = None
        self.linear_rounds = 0
        self._flat_cache = None

    # -- core info ---------------------------------------------------------
    @property
    def objective(self) -> str:
        return self.params.get("objective", "reg:squarederror")

    @property
    def num_class(self) -> int:
        return int(self.params.get("num_class", 0) or 0)

    @property
    def num_features(self) -> int:
        return int(self.params.get("num_feature", 0) or 0)

    @property
    def base_score(self) -> float:
        bs = self.params.get("base_score", None)
        if bs is None:
            return 0.5
        return float(bs)

    @property
    def num_parallel_tree(self) -> int:
        return max(1, int(self.params.get("num_parallel_tree", 1) or 1))

    def num_boosted_rounds(self) -> int:
        if self.params.get("booster") == "gblinear":
            return int(self.linear_rounds)
        k = max(1, self.num_class) * self.num_parallel_tree
        return len(self.trees) // k

    @property
    def best_ntree_limit(self) -> int:
        """Legacy xgboost attribute: trees up to the best iteration."""
        it = (
            self.best_iteration
            if self.best_iteration is not None
            else self.num_boosted_rounds() - 1
        )
        return (it + 1) * max(1, self.num_class) * self.num_parallel_tree

    def append_round(self, trees: Sequence[Tree], classes: Sequence[int]):
        self.trees.extend(trees)
        self.tree_info.extend(int(c) for c in classes)
        self._flat_cache = None

    # -- attributes (xgboost API parity) -----------------------------------
    def attr(self, key):
        return self.attributes_.get(key)

    def set_attr(self, **kwargs):
        for k, v in kwargs.items():
            if v is None:
                self.attributes_.pop(k, None)
            else:
                self.attributes_[k] = str(v)

    def attributes(self):
        return dict(self.attributes_)

    # -- prediction --------------------------------------------------------
    def _flat_trees(self, device="cpu"):
        if self._flat_cache is not None and self._flat_cache[0] == str(device):
            return self._flat_cache[1]
        ptr = [0]
        feats, thrs, lefts, dls, vals = [], [], [], [], []
        for t in self.trees:
            feats.append(t.feat)
            thrs.append(t.thr)
            lefts.append(t.left)
            dls.append(t.default_left)
            vals.append(t.value)
            ptr.append(ptr[-1] + t.num_nodes)
        if not self.trees:
            feats = [np.zeros(0, np.int32)]
            thrs = [np.zeros(0, np.float32)]
            lefts = [np.zeros(0, np.int32)]
            dls = [np.zeros(0, np.uint8)]
            vals = [np.zeros(0, np.float32)]
        flat = {
            "feat": torch.from_numpy(np.concatenate(feats)).to(device),
            "thr": torch.from_numpy(np.concatenate(thrs)).to(device),
            "left": torch.from_numpy(np.concatenate(lefts)).to(device),
            "default_left": torch.from_numpy(np.concatenate(dls)).to(device),
            "value": torch.from_numpy(np.concatenate(vals)).to(device),
            "tree_ptr": torch.tensor(ptr, dtype=torch.int32, device=device),
            "tree_info": torch.tensor(
                self.tree_info or [0], dtype=torch.int32, device=device
            ),
        }
        self._flat_cache = (str(device), flat)
        return flat

    def predict_margin_tensor(
        self, X: torch.Tensor, iteration_range=None
    ) -> torch.Tensor:
        """Raw margin per row (``[n]`` or ``[n, num_class]``)."""
        from xgboost_ray_amd import ops
        from xgboost_ray_amd.engine.objectives import get_objective

        obj = get_objective(self.objective, self.num_class)
        n = X.shape[0]
        k = max(1, self.num_class)
        base = obj.prob_to_margin(self.base_score)
        if self.params.get("booster") == "gblinear":
            # linear model: margin = X @ w + bias (+ base); iteration_range
            # has no meaning for a single weight vector
            w = torch.from_numpy(
                np.asarray(self.linear_weights, dtype=np.float32)
            ).to(X.device)
            m = X @ w[:-1] + w[-1] + base
            return m[:, 0] if k == 1 else m
        flat = self._flat_trees(X.device)

        lo, hi = 0, self.num_boosted_rounds()
        if iteration_range is not None:
            lo, hi = iteration_range
            hi = min(hi, self.num_boosted_rounds()) if hi else self.num_boosted_rounds()
        per_round = k * self.num_parallel_tree

        outs = []
        for cls in range(k):
            out = torch.full((n,), base, dtype=torch.float32, device=X.device)
            sel = [
                i
                for i in range(len(self.trees))
                if self.tree_info[i] == cls and lo <= i // per_round < hi
            ]
            if sel:
                # contiguous runs share ptr structure; do it per selected tree
                # group to keep the op simple
                ptr = flat["tree_ptr"]
                sub_ptr = [0]
                sub_arrays = {kk: [] for kk in ("feat", "thr", "left", "default_left", "value")}
                for i in sel:
                    s, e = int(ptr[i]), int(ptr[i + 1])
                    for kk in sub_arrays:
                        sub_arrays[kk].append(flat[kk][s:e])
                    sub_ptr.append(sub_ptr[-1] + (e - s))
                ops.predict_trees(
                    X,
                    torch.cat(sub_arrays["feat"]),
                    torch.cat(sub_arrays["thr"]),
                    torch.cat(sub_arrays["left"]),
                    torch.cat(sub_arrays["default_left"]),
                    torch.cat(sub_arrays["value"]),
                    torch.tensor(sub_ptr, dtype=torch.int32, device=X.device),
                    out,
                )
            outs.append(out)
        if k == 1:
            return outs[0]
        return torch.stack(outs, dim=1)

    def predict(
        self,
        data: Union[DMatrix, np.ndarray],
        output_margin: bool = False,
        iteration_range=None,
        validate_features: bool = True,
        pred_leaf: bool = False,
        pred_contribs: bool = False,
        pred_interactions: bool = False,
        **kwargs,
    ) -> np.ndarray:
        from xgboost_ray_amd.engine.objectives import get_objective

        if isinstance(data, DMatrix):
            X = data.data
            bm = data.base_margin
            in_names = data.feature_names
        else:
            in_names = (
                [str(c) for c in data.columns]
                if hasattr(data, "columns") else None
            )
            X = _as_float32_matrix(data)
            bm = None
        if validate_features and self.feature_names and in_names:
            if list(in_names) != list(self.feature_names):
                # xgboost semantics: named inputs must match the names
                # the model was trained with, in order
                raise ValueError(
                    "feature_names mismatch: model expects "
                    f"{self.feature_names}, got {in_names}"
                )
        if pred_leaf:
            return self.predict_leaf(X, iteration_range)
        if pred_contribs:
            if kwargs.get("approx_contribs"):
                return self.predict_contribs_approx(X, iteration_range)
            return self.predict_contribs(X, iteration_range)
        if pred_interactions:
            return self.predict_interactions(X, iteration_range)
        Xt = torch.from_numpy(np.ascontiguousarray(X, dtype=np.float32))
        # MI355X-first routing: when a GPU is present, bulk prediction
        # runs the LDS tree-walk kernel there (the CPU level-walk is for
        # GPU-less test environments; at 5M rows x 50 trees it is
        # minutes vs 30 ms on device)
        if (
            not Xt.is_cuda
            and Xt.shape[0] >= 16384
            and torch.cuda.is_available()
        ):
            Xt = Xt.cuda()
        margin = self.predict_margin_tensor(Xt, iteration_range)
        if bm is not None:
            margin = margin + torch.from_numpy(bm).to(margin.dtype).reshape(
                margin.shape[0], *([1] * (margin.dim() - 1))
            )
        if output_margin:
            return margin.cpu().numpy()
        obj = get_objective(self.objective, self.num_class)
        return obj.transform_prediction(margin).cpu().numpy()


    def predict_contribs_approx(self, X, iteration_range=None) -> np.ndarray:
        """Saabas-style approximate contributions (xgboost
        ``pred_contribs`` with ``approx_contribs=True``): each split on
        the walked root-to-leaf path credits its feature with the change
        in the cover-weighted expected value. Fully vectorized level
        walk - orders of magnitude faster than exact TreeSHAP, satisfies
        the same additivity identity contribs.sum(-1) == margin."""
        from xgboost_ray_amd.engine.objectives import get_objective

        X = _as_float32_matrix(X)
        n, F = X.shape
        lo, hi = 0, self.num_boosted_rounds()
        if iteration_range is not None:
            lo, hi = iteration_range
            hi = hi or self.num_boosted_rounds()
        obj = get_objective(self.objective, self.num_class)
        base = float(obj.prob_to_margin(self.base_score))
        k_cls = max(1, self.num_class)
        k = k_cls * self.num_parallel_tree
        if self.num_class > 1:
            out = np.zeros((n, self.num_class, F + 1), np.float64)
        else:
            out = np.zeros((n, F + 1), np.float64)
        out[..., F] += base  # global intercept joins the bias column
        rows = np.arange(n)
        for ti in range(lo * k, min(hi * k, len(self.trees))):
            t = self.trees[ti]
            cls = self.tree_info[ti] if self.num_class > 1 else 0
            dest = out[:, cls, :] if self.num_class > 1 else out
            mean_val = np.zeros(t.num_nodes, np.float64)
            cover = t.cover.astype(np.float64)
            _fill_node_means(t, mean_val, cover)
            dest[:, F] += mean_val[0]  # bias: tree expected value
            cur = np.zeros(n, np.int64)
            while True:
                f = t.feat[cur]
                inner = f >= 0
                if not inner.any():
                    break
                r = rows[inner]
                fr = f[inner]
                cr = cur[inner]
                x = X[r, fr]
                left = t.left[cr]
                go_left = x < t.thr[cr]
                go_left = np.where(
                    np.isnan(x), t.default_left[cr].astype(bool), go_left
                )
                child = np.where(go_left, left, left + 1)
                # credit the split feature with the expectation change
                np.add.at(dest, (r, fr), mean_val[child] - mean_val[cr])
                nxt = cur.copy()
                nxt[inner] = child
                cur = nxt
        return out

    def predict_contribs(self, X, iteration_range=None) -> np.ndarray:
        """Exact TreeSHAP feature attributions (xgboost ``pred_contribs``;
        the reference forwards predict kwargs to xgboost, main.py:795-810).

        Returns [n, F+1] (multi-class: [n, num_class, F+1]); the last
        column is the bias (expected value). Rows satisfy additivity:
        contribs.sum(-1) == margin. Implements the Lundberg et al.
        polynomial-time EXTEND/UNWIND recursion per tree."""
        X = _as_float32_matrix(X)
        n, F = X.shape
        lo, hi = 0, self.num_boosted_rounds()
        if iteration_range is not None:
            lo, hi = iteration_range
            hi = hi or self.num_boosted_rounds()
        if self.num_class > 1:
            from xgboost_ray_amd.engine.objectives import get_objective

            obj = get_objective(self.objective, self.num_class)
            base = float(obj.prob_to_margin(self.base_score))
            outc = np.full(
                (n, self.num_class, F + 1), 0.0, dtype=np.float64
            )
            k = max(1, self.num_class) * self.num_parallel_tree
            for ti in range(lo * k, min(hi * k, len(self.trees))):
                cls = self.tree_info[ti]
                t = self.trees[ti]
                self._shap_one_tree(
                    t, X, outc[:, cls, :], 1.0 / self.num_parallel_tree
                    if self.num_parallel_tree > 1 else 1.0,
                )
            outc[:, :, F] += base
            return outc
        out = np.zeros((n, F + 1), dtype=np.float64)
        k = self.num_parallel_tree
        trees = self.trees[lo * k : hi * k]
        scale = 1.0 / k if k > 1 else 1.0
        for t in trees:
            self._shap_one_tree(t, X, out, scale)
        # bias column: margin expectation = sum of tree means (added in
        # _tree_shap) + the model's base margin
        from xgboost_ray_amd.engine.objectives import get_objective

        obj = get_objective(self.objective, self.num_class)
        out[:, F] += float(obj.prob_to_margin(self.base_score))
        return out


    @staticmethod
    def _shap_one_tree(t, X, out, scale):
        """Accumulate one tree's SHAP contributions into out [n, F+1]."""
        mean_val = np.zeros(t.num_nodes)
        cover = t.cover.astype(np.float64)
        _fill_node_means(t, mean_val, cover)
        for i in range(X.shape[0]):
            _tree_shap(t, X[i], mean_val, cover, out[i], scale)

    def predict_interactions(self, X, iteration_range=None) -> np.ndarray:
        """SHAP interaction values (xgboost ``pred_interactions``).

        [n, F+1, F+1]: off-diagonal [i, j] is the interaction between
        features i and j ((shap_j | i present) - (shap_j | i absent))/2,
        the diagonal holds main effects so each row i sums to the
        feature's SHAP value, and [F, F] is the bias - the whole matrix
        sums to the margin. Single-class models; O(F) TreeSHAP passes
        per row, so intended for explanation-sized batches."""
        if self.num_class > 1:
            raise NotImplementedError(
                "pred_interactions for multi-class models is not supported"
            )
        X = _as_float32_matrix(X)
        n, F = X.shape
        contribs = self.predict_contribs(X, iteration_range)
        lo, hi = 0, self.num_boosted_rounds()
        if iteration_range is not None:
            lo, hi = iteration_range
            hi = hi or self.num_boosted_rounds()
        k = self.num_parallel_tree
        trees = self.trees[lo * k : hi * k]
        scale = 1.0 / k if k > 1 else 1.0
        M = np.zeros((n, F + 1, F + 1), dtype=np.float64)
        on = np.zeros((n, F + 1), dtype=np.float64)
        off = np.zeros((n, F + 1), dtype=np.float64)
        for i in range(F):
            on[:] = 0.0
            off[:] = 0.0
            for t in trees:
                nn = t.num_nodes
                mean_val = np.zeros(nn)
                cover = t.cover.astype(np.float64)
                _fill_node_means(t, mean_val, cover)
                for r in range(n):
                    _tree_shap(t, X[r], mean_val, cover, on[r], scale,
                               condition=1, condition_feature=i)
                    _tree_shap(t, X[r], mean_val, cover, off[r], scale,
                               condition=-1, condition_feature=i)
            M[:, i, :F] = (on[:, :F] - off[:, :F]) / 2.0
            M[:, i, i] = 0.0
        # symmetrize then set diagonals so each row sums to the SHAP value
        M[:, :F, :F] = (M[:, :F, :F] + np.transpose(M[:, :F, :F],
                                                    (0, 2, 1))) / 2.0
        for i in range(F):
            M[:, i, i] = contribs[:, i] - M[:, i, :F].sum(axis=1)
        M[:, F, F] = contribs[:, F]
        return M

    def predict_leaf(self, X, iteration_range=None) -> np.ndarray:
        """Leaf index per (row, tree) - xgboost ``pred_leaf=True``."""
        X = _as_float32_matrix(X)
        lo, hi = 0, self.num_boosted_rounds()
        if iteration_range is not None:
            lo, hi = iteration_range
            hi = hi or self.num_boosted_rounds()
        k = max(1, self.num_class) * self.num_parallel_tree
        trees = self.trees[lo * k : hi * k]
        n = X.shape[0]
        out = np.zeros((n, len(trees)), dtype=np.int32)
        for ti, t in enumerate(trees):
            cur = np.zeros(n, dtype=np.int32)
            # walk all rows level-synchronously (vectorized numpy)
            while True:
                f = t.feat[cur]
                inner = f >= 0
                if not inner.any():
                    break
                rows = np.nonzero(inner)[0]
                fv = X[rows, f[rows]]
                left = t.left[cur[rows]]
                go_left = fv < t.thr[cur[rows]]
                miss = np.isnan(fv)
                go_left = np.where(
                    miss, t.default_left[cur[rows]].astype(bool), go_left
                )
                cur[rows] = np.where(go_left, left, left + 1)
            out[:, ti] = cur
        return out

    # -- persistence -------------------------------------------------------
    def __getstate__(self):
        state = self.__dict__.copy()
        state["_flat_cache"] = None
        return state

    def __setstate__(self, state):
        self.__dict__.update(state)

    def save_model(self, fname: str):
        """Persist in the XGBoost model format the extension selects:
        ``.json`` -> JSON, ``.ubj``/``.xgb`` -> UBJSON (the binary format
        stock XGBoost >= 1.6 dispatches ``.ubj`` to with no content
        sniffing; the reference's user contract saves ``model.xgb``,
        reference README.md:78 and xgboost_ray/tune.py:130-156)."""
        name = str(fname)
        if name.endswith(".ubj") or name.endswith(".xgb"):
            with open(fname, "wb") as f:
                f.write(ubjson.dumps(self._to_json_dict()))
        else:
            with open(fname, "w") as f:
                json.dump(self._to_json_dict(), f)

    def save_config(self) -> str:
        """Learner configuration as a JSON string (xgboost
        Booster.save_config surface: enough for config snapshots and
        load_config round-trips; the model itself uses save_model)."""
        return json.dumps({
            "learner": {
                "generic_param": {},
                "gradient_booster": {
                    "name": self.params.get("booster", "gbtree"),
                },
                "learner_model_param": {
                    "base_score": repr(self.base_score),
                    "num_class": str(self.num_class),
                    "num_feature": str(self.num_features),
                },
                "learner_train_param": {
                    "objective": self.objective,
                },
                "params": {
                    k: v for k, v in self.params.items()
                    if isinstance(v, (str, int, float, bool))
                },
            }
        })

    def load_config(self, config: str):
        doc = json.loads(config)
        learner = doc.get("learner", {})
        lp = learner.get("learner_model_param", {})
        for key in ("base_score", "num_class", "num_feature"):
            if key in lp:
                self.params[key] = lp[key]
        if "learner_train_param" in learner:
            obj = learner["learner_train_param"].get("objective")
            if obj:
                self.params["objective"] = obj
        self.params.update(learner.get("params", {}))
        return self

    def save_raw(self, raw_format: str = "json") -> bytes:
        if raw_format == "ubj":
            return ubjson.dumps(self._to_json_dict())
        if raw_format == "json":
            return json.dumps(self._to_json_dict()).encode()
        raise ValueError(
            f"raw_format must be 'json' or 'ubj', got {raw_format!r}"
        )

    @staticmethod
    def _parse_model_bytes(raw: bytes):
        """Content-sniffed parse: both formats open with ``{`` (0x7b), but
        in UBJSON the next byte is an integer-length marker for the first
        key (or a ``$``/``#`` container header) while JSON follows with
        whitespace, ``"`` or ``}``."""
        if raw[:1] == b"{" and raw[1:2] in b"iUIlL$#N":
            return ubjson.loads(raw)
        return json.loads(raw.decode())

    def load_model(self, fname):
        if isinstance(fname, (bytes, bytearray)):
            raw = bytes(fname)
        else:
            with open(fname, "rb") as f:
                raw = f.read()
        self._from_json_dict(self._parse_model_bytes(raw))
        return self

    def _to_json_dict(self) -> Dict:
        """XGBoost JSON model schema (xgboost doc: model format v2/v3)."""
        learner_param = {
            "base_score": repr(self.base_score),
            "boost_from_average": "1",
            "num_class": str(self.num_class),
            "num_feature": str(self.num_features),
            "num_target": "1",
        }
        trees_json = []
        for tid, t in enumerate(self.trees):
            n = t.num_nodes
            left = t.left.astype(np.int64)
            right = np.where(left >= 0, left + 1, -1)
            parent = (
                t.parent
                if t.parent is not None
                else _parents_from_children(t.left)
            )
            is_leaf = t.feat < 0
            trees_json.append(
                {
                    "base_weights": t.value.astype(float).tolist(),
                    "categories": [],
                    "categories_nodes": [],
                    "categories_segments": [],
                    "categories_sizes": [],
                    "default_left": t.default_left.astype(int).tolist(),
                    "id": tid,
                    "left_children": left.astype(int).tolist(),
                    "right_children": right.astype(int).tolist(),
                    "loss_changes": t.gain.astype(float).tolist(),
                    "parents": np.asarray(parent).astype(int).tolist(),
                    "split_conditions": np.where(
                        is_leaf, t.value, t.thr
                    ).astype(float).tolist(),
                    "split_indices": np.maximum(t.feat, 0).astype(int).tolist(),
                    "split_type": [0] * n,
                    "sum_hessian": t.cover.astype(float).tolist(),
                    "tree_param": {
                        "num_deleted": "0",
                        "num_feature": str(self.num_features),
                        "num_nodes": str(n),
                        "size_leaf_vector": "1",
                    },
                }
            )
        num_rounds = self.num_boosted_rounds()
        k = max(1, self.num_class) * self.num_parallel_tree
        if self.params.get("booster") == "gblinear":
            # xgboost gblinear schema: flat weights, bias block last
            gb = {
                "model": {
                    "weights": np.asarray(self.linear_weights)
                    .astype(float).reshape(-1).tolist(),
                },
                "name": "gblinear",
            }
        else:
            gb = {
                "model": {
                    "gbtree_model_param": {
                        "num_trees": str(len(self.trees)),
                        "num_parallel_tree": str(self.num_parallel_tree),
                    },
                    "iteration_indptr": [i * k for i in range(num_rounds + 1)],
                    "tree_info": list(self.tree_info),
                    "trees": trees_json,
                },
                "name": "gbtree",
            }
        doc = {
            "learner": {
                "attributes": dict(self.attributes_),
                "feature_names": self.feature_names or [],
                "feature_types": [],
                "gradient_booster": gb,
                "learner_model_param": learner_param,
                "objective": _objective_json(self.objective),
            },
            "version": [2, 1, 0],
        }
        if self.params.get("booster") == "gblinear":
            doc["learner"]["attributes"]["gblinear_rounds"] = str(
                self.linear_rounds
            )
        return doc

    def _from_json_dict(self, doc: Dict):
        learner = doc["learner"]
        lp = learner["learner_model_param"]
        self.params["base_score"] = float(lp.get("base_score", 0.5))
        self.params["num_class"] = int(lp.get("num_class", 0))
        self.params["num_feature"] = int(lp.get("num_feature", 0))
        self.params["objective"] = learner.get("objective", {}).get(
            "name", "reg:squarederror"
        )
        self.feature_names = learner.get("feature_names") or None
        self.attributes_ = dict(learner.get("attributes", {}))
        model = learner["gradient_booster"]["model"]
        if learner["gradient_booster"].get("name") == "gblinear":
            self.params["booster"] = "gblinear"
            F = int(lp.get("num_feature", 0))
            k = max(1, int(lp.get("num_class", 0) or 0) or 1)
            w = np.asarray(model["weights"], dtype=np.float64)
            self.linear_weights = w.reshape(F + 1, k)
            self.linear_rounds = int(
                self.attributes_.pop("gblinear_rounds", 0) or 0
            )
            self.trees = []
            self.tree_info = []
            self._flat_cache = None
            return
        gbm_param = model.get("gbtree_model_param", {})
        if "num_parallel_tree" in gbm_param:
            self.params["num_parallel_tree"] = int(
                gbm_param["num_parallel_tree"]
            )
        self.tree_info = [int(x) for x in model.get("tree_info", [])]
        self.trees = []
        for tj in model["trees"]:
            n = int(tj["tree_param"]["num_nodes"])
            left = np.asarray(tj["left_children"], dtype=np.int32)
            is_leaf = left < 0
            feat = np.where(
                is_leaf, -1, np.asarray(tj["split_indices"], dtype=np.int32)
            ).astype(np.int32)
            cond = np.asarray(tj["split_conditions"], dtype=np.float32)
            value = np.where(is_leaf, cond, np.asarray(tj["base_weights"], np.float32)).astype(np.float32)
            self.trees.append(
                Tree(
                    feat=feat,
                    thr=np.where(is_leaf, 0, cond).astype(np.float32),
                    left=left,
                    default_left=np.asarray(tj["default_left"], np.uint8),
                    value=value,
                    gain=np.asarray(tj["loss_changes"], np.float32),
                    cover=np.asarray(tj["sum_hessian"], np.float32),
                    parent=np.asarray(tj["parents"], np.int32),
                )
            )
        if not self.tree_info:
            self.tree_info = [0] * len(self.trees)
        self._flat_cache = None

    def get_score(self, fmap: str = "", importance_type: str = "weight"):
        """Per-feature importance (xgboost Booster.get_score parity).

        importance_type: weight (split count), gain, total_gain, cover,
        total_cover.
        """
        counts: Dict[str, float] = {}
        gains: Dict[str, float] = {}
        covers: Dict[str, float] = {}

        def fname(f):
            if self.feature_names and f < len(self.feature_names):
                return self.feature_names[f]
            return f"f{f}"

        for t in self.trees:
            for nid in range(t.num_nodes):
                f = int(t.feat[nid])
                if f < 0:
                    continue
                key = fname(f)
                counts[key] = counts.get(key, 0.0) + 1.0
                gains[key] = gains.get(key, 0.0) + float(t.gain[nid])
                covers[key] = covers.get(key, 0.0) + float(t.cover[nid])
        if importance_type == "weight":
            return counts
        if importance_type == "total_gain":
            return gains
        if importance_type == "gain":
            return {k: v / counts[k] for k, v in gains.items()}
        if importance_type == "total_cover":
            return covers
        if importance_type == "cover":
            return {k: v / counts[k] for k, v in covers.items()}
        raise ValueError(f"Unknown importance_type: {importance_type}")

    def trees_to_dataframe(self, fmap=""):
        """Flat per-node table (xgboost Booster.trees_to_dataframe parity):
        columns Tree/Node/ID/Feature/Split/Yes/No/Missing/Gain/Cover."""
        import pandas as pd

        rows = []
        for ti, t in enumerate(self.trees):
            for nid in range(t.num_nodes):
                f = int(t.feat[nid])
                leaf = f < 0
                if leaf:
                    feature, split, yes, no, miss = "Leaf", None, None, None, None
                    gain = float(t.value[nid])
                else:
                    if self.feature_names and f < len(self.feature_names):
                        feature = self.feature_names[f]
                    else:
                        feature = f"f{f}"
                    split = float(t.thr[nid])
                    l = int(t.left[nid])
                    yes, no = f"{ti}-{l}", f"{ti}-{l + 1}"
                    miss = yes if t.default_left[nid] else no
                    gain = float(t.gain[nid])
                rows.append(
                    {
                        "Tree": ti,
                        "Node": nid,
                        "ID": f"{ti}-{nid}",
                        "Feature": feature,
                        "Split": split,
                        "Yes": yes,
                        "No": no,
                        "Missing": miss,
                        "Gain": gain,
                        "Cover": float(t.cover[nid]),
                    }
                )
        return pd.DataFrame(rows)

    def get_dump(self, fmap="", with_stats=False, dump_format="text"):
        out = []
        for t in self.trees:
            lines = []

            def rec(nid, depth):
                ind = "\t" * depth
                if t.feat[nid] < 0:
                    lines.append(f"{ind}{nid}:leaf={t.value[nid]:.9g}")
                else:
                    f = int(t.feat[nid])
                    yes, no = int(t.left[nid]), int(t.left[nid]) + 1
                    miss = yes if t.default_left[nid] else no
                    lines.append(
                        f"{ind}{nid}:[f{f}<{t.thr[nid]:.9g}] "
                        f"yes={yes},no={no},missing={miss}"
                    )
                    rec(yes, depth + 1)
                    rec(no, depth + 1)

            rec(0, 0)
            out.append("\n".join(lines) + "\n")
        return out


def _parents_from_children(left: np.ndarray) -> np.ndarray:
    parent = np.full(len(left), -1, dtype=np.int32)
    # match XGBoost convention: root parent = 2147483647
    parent[0] = 2147483647
    for i, l in enumerate(left):
        if l >= 0:
            parent[l] = i
            parent[l + 1] = i
    return parent


def _objective_json(name: str) -> Dict:
    if name.startswith("binary:"):
        return {"name": name, "reg_loss_param": {"scale_pos_weight": "1"}}
    if name.startswith("multi:"):
        return {"name": name, "softmax_multiclass_param": {"num_class": "0"}}
    if name.startswith("rank:"):
        return {"name": name, "lambdarank_param": {}}
    return {"name": name, "reg_loss_param": {"scale_pos_weight": "1"}}
