"""Objective functions: per-row gradient/hessian computation.

MI355X-native equivalent of XGBoost's C++/CUDA objective registry
(reference delegates these to libxgboost; SURVEY.md #2.3 'Objectives &
metrics'). All math runs on torch tensors so the same code executes on CPU
(tests) and on GPU (fused elementwise kernels take over for the hot ones).
Custom Python objectives (callables) stay supported, matching the
reference's passthrough (reference test_xgboost_api.py:77-152).
"""

import math
from typing import Optional

import torch

_EPS = 1e-16


def sigmoid_sizeinv(t: torch.Tensor) -> torch.Tensor:
    """``torch.sigmoid`` whose CPU results do not depend on array length.

    torch's CPU kernel runs a sleef vector body plus a scalar tail whose
    results can differ by 1 ulp for the same input value; a row that sits
    in the tail at one shard size and in the body at another broke the
    bitwise distributed==single invariant (7 quantized-gradient flips on
    a 4000-row matrix split 2 ways). Padding to a vector-width multiple
    keeps every real element in the body. GPU elementwise kernels are
    position-independent, so no padding there.
    """
    return elemwise_sizeinv(torch.sigmoid, t)


def elemwise_sizeinv(fn, t: torch.Tensor) -> torch.Tensor:
    """Apply an elementwise transcendental with length-invariant CPU
    results (see sigmoid_sizeinv) - used for sigmoid/exp/log1p in
    gradient paths that must be bitwise world-size invariant."""
    flat = t.reshape(-1)
    if t.device.type != "cpu" or flat.numel() % 64 == 0:
        return fn(t)
    pad = (-flat.numel()) % 64
    tp = torch.cat([flat, torch.zeros(pad, dtype=t.dtype)])
    return fn(tp)[: flat.numel()].reshape(t.shape)


class Objective:
    name = "base"
    n_class = 0
    default_metric = "rmse"

    def prob_to_margin(self, base_score: float) -> float:
        return base_score

    def gradients(
        self,
        margin: torch.Tensor,
        label: torch.Tensor,
        weight: Optional[torch.Tensor] = None,
        qid: Optional[torch.Tensor] = None,
    ) -> torch.Tensor:
        """Return fp32 gpair [n, 2] (or [n, k, 2] for multiclass)."""
        raise NotImplementedError

    def transform_prediction(self, margin: torch.Tensor) -> torch.Tensor:
        return margin

    def validate_label(self, label: torch.Tensor):
        pass

    def _apply_weight(self, g, h, weight):
        if weight is not None:
            g = g * weight
            h = h * weight
        return torch.stack([g, h], dim=-1)


class SquaredError(Objective):
    name = "reg:squarederror"
    default_metric = "rmse"

    def gradients(self, margin, label, weight=None, qid=None):
        if margin.is_cuda and margin.dim() == 1:
            from xgboost_ray_amd import ops

            out = ops.grad_fused(margin, label, weight, 1.0, mode=0)
            if out is not None:
                gp, mx = out
                gp._rxgb_absmax = mx  # consumed by _quantize
                return gp
        g = margin - label
        h = torch.ones_like(margin)
        return self._apply_weight(g, h, weight)


class AbsoluteError(Objective):
    name = "reg:absoluteerror"
    default_metric = "mae"

    def gradients(self, margin, label, weight=None, qid=None):
        g = torch.sign(margin - label)
        h = torch.ones_like(margin)
        return self._apply_weight(g, h, weight)


class Logistic(Objective):
    """binary:logistic - probability output."""

    name = "binary:logistic"
    default_metric = "logloss"

    def __init__(self, scale_pos_weight: float = 1.0):
        self.scale_pos_weight = float(scale_pos_weight)

    def prob_to_margin(self, base_score):
        base_score = min(max(base_score, _EPS), 1.0 - _EPS)
        return -math.log(1.0 / base_score - 1.0)

    def gradients(self, margin, label, weight=None, qid=None):
        if margin.is_cuda and margin.dim() == 1:
            from xgboost_ray_amd import ops

            out = ops.grad_fused(
                margin, label, weight, self.scale_pos_weight, mode=1
            )
            if out is not None:
                gp, mx = out
                gp._rxgb_absmax = mx  # consumed by _quantize
                return gp
        p = sigmoid_sizeinv(margin)
        g = p - label
        h = torch.clamp(p * (1.0 - p), min=_EPS)
        if self.scale_pos_weight != 1.0:
            # XGBoost semantics: positive rows weighted by scale_pos_weight
            w = 1.0 + (self.scale_pos_weight - 1.0) * label
            g = g * w
            h = h * w
        return self._apply_weight(g, h, weight)

    def transform_prediction(self, margin):
        return sigmoid_sizeinv(margin)

    def validate_label(self, label):
        if bool((label < 0).any()) or bool((label > 1).any()):
            raise ValueError("label must be in [0,1] for binary:logistic")


class LogisticRaw(Logistic):
    """binary:logitraw - margin output."""

    name = "binary:logitraw"
    default_metric = "logloss"

    def transform_prediction(self, margin):
        return margin


class RegLogistic(Logistic):
    name = "reg:logistic"
    default_metric = "rmse"


class SoftmaxBase(Objective):
    default_metric = "mlogloss"

    def __init__(self, n_class: int):
        if n_class < 2:
            raise ValueError("multi objectives need num_class >= 2")
        self.n_class = n_class

    def gradients(self, margin, label, weight=None, qid=None):
        # margin: [n, k]
        p = torch.softmax(margin, dim=1)
        y = torch.nn.functional.one_hot(label.long(), self.n_class).to(p.dtype)
        g = p - y
        h = torch.clamp(2.0 * p * (1.0 - p), min=_EPS)
        if weight is not None:
            g = g * weight.unsqueeze(1)
            h = h * weight.unsqueeze(1)
        return torch.stack([g, h], dim=-1)  # [n, k, 2]

    def validate_label(self, label):
        mx = int(label.max()) if label.numel() else 0
        if mx >= self.n_class:
            raise ValueError(
                f"label {mx} >= num_class {self.n_class}"
            )


class SoftmaxClass(SoftmaxBase):
    name = "multi:softmax"
    default_metric = "merror"

    def transform_prediction(self, margin):
        return margin.argmax(dim=1).to(torch.float32)


class SoftProb(SoftmaxBase):
    name = "multi:softprob"
    default_metric = "mlogloss"

    def transform_prediction(self, margin):
        return torch.softmax(margin, dim=1)


class LambdaRankBase(Objective):
    """Pairwise LambdaRank objectives over qid groups.

    Deterministic all-pairs formulation (pairs of rows with different
    relevance within a query group); NDCG variant weights each pair by its
    |delta-NDCG|. Reference semantics: rank:pairwise / rank:ndcg passed
    through to libxgboost (reference test_end_to_end.py:374-424).
    """

    ndcg_weighting = False
    default_metric = "map"

    def gradients(self, margin, label, weight=None, qid=None):
        if qid is None:
            raise ValueError("rank objectives require qid/group information")
        if margin.device.type == "cuda":
            return self._gradients_gpu(margin, label, weight, qid)
        return self._gradients_generic(margin, label, weight, qid)

    def _gradients_generic(self, margin, label, weight=None, qid=None):
        device = margin.device
        g = torch.zeros_like(margin)
        h = torch.zeros_like(margin)
        qid = qid.to(device)
        # group boundaries (qid must be sorted)
        change = torch.ones_like(qid, dtype=torch.bool)
        change[1:] = qid[1:] != qid[:-1]
        starts = torch.nonzero(change).flatten().tolist() + [len(qid)]
        for gi in range(len(starts) - 1):
            s, e = starts[gi], starts[gi + 1]
            if e - s < 2:
                continue
            m = margin[s:e]
            y = label[s:e]
            diff_y = y.unsqueeze(1) - y.unsqueeze(0)  # [g, g]
            pos = diff_y > 0  # i more relevant than j
            if not bool(pos.any()):
                continue
            sij = m.unsqueeze(1) - m.unsqueeze(0)
            rho = torch.sigmoid(-sij)  # d/ds_i of log-loss for pair (i,j)
            lam = rho
            hess = torch.clamp(rho * (1.0 - rho), min=_EPS)
            if getattr(self, "map_weighting", False):
                # |delta AP| for swapping i, j at the current ranking
                # (binary relevance; closed form via 1/k prefix sums)
                rel = (y > 0).to(m.dtype)
                R = rel.sum()
                if float(R) <= 0:
                    continue
                order = torch.argsort(m, descending=True)
                rank = torch.empty_like(order)
                rank[order] = torch.arange(e - s, device=device)
                rel_sorted = rel[order]
                prefix = torch.cumsum(rel_sorted, 0)  # rel count in top k+1
                invk = 1.0 / torch.arange(
                    1, e - s + 1, device=device, dtype=m.dtype
                )
                S = torch.cumsum(rel_sorted * invk, 0)  # sum 1/k over rel
                # per-doc terms at its own rank (0-based rank r -> k=r+1)
                term = prefix[rank] * invk[rank]       # prefix[k]/k
                Sr = S[rank]                           # S up to own rank
                # pair (i above j): delta = |term_j' - term_i - (S_{j-1}-S_i)|
                ra = rank.unsqueeze(1)
                rb = rank.unsqueeze(0)
                lo_r = torch.minimum(ra, rb)
                hi_r = torch.maximum(ra, rb)
                S_lo = S[lo_r]
                S_hi_m1 = torch.where(
                    hi_r > 0, S[torch.clamp(hi_r - 1, min=0)],
                    torch.zeros_like(S_lo),
                )
                t_lo = (prefix[lo_r]) * invk[lo_r]
                t_hi = (prefix[hi_r]) * invk[hi_r]
                delta = torch.abs(t_hi - t_lo - (S_hi_m1 - S_lo)) / R
                lam = lam * delta
                hess = hess * delta
            elif self.ndcg_weighting:
                # |delta NDCG| for swapping i, j at current ranking
                order = torch.argsort(m, descending=True)
                rank = torch.empty_like(order)
                rank[order] = torch.arange(e - s, device=device)
                gain = torch.pow(2.0, y) - 1.0
                disc = 1.0 / torch.log2(rank.to(m.dtype) + 2.0)
                ideal_gain, _ = torch.sort(gain, descending=True)
                ideal_disc = 1.0 / torch.log2(
                    torch.arange(e - s, device=device, dtype=m.dtype) + 2.0
                )
                idcg = (ideal_gain * ideal_disc).sum()
                if idcg <= 0:
                    continue
                dg = gain.unsqueeze(1) - gain.unsqueeze(0)
                dd = disc.unsqueeze(1) - disc.unsqueeze(0)
                delta = torch.abs(dg * dd) / idcg
                lam = lam * delta
                hess = hess * delta
            lam = torch.where(pos, lam, torch.zeros_like(lam))
            hess = torch.where(pos, hess, torch.zeros_like(hess))
            # pair (i, j): i should rank above j -> push m_i up, m_j down
            g[s:e] += -lam.sum(dim=1) + lam.sum(dim=0)
            h[s:e] += hess.sum(dim=1) + hess.sum(dim=0)
        return self._apply_weight(g, h, weight)

    def _gradients_gpu(self, margin, label, weight, qid):
        """HIP kernel path: one workgroup per query group, deterministic
        per-doc accumulation (no atomics)."""
        from xgboost_ray_amd.ops import gpu as gpu_ops

        device = margin.device
        qid = qid.to(device)
        n = margin.shape[0]
        change = torch.ones(n, dtype=torch.bool, device=device)
        change[1:] = qid[1:] != qid[:-1]
        group_id = torch.cumsum(change.long(), 0) - 1  # [n]
        counts = torch.bincount(group_id)
        group_ptr = torch.zeros(
            len(counts) + 1, dtype=torch.int64, device=device
        )
        torch.cumsum(counts, 0, out=group_ptr[1:])

        if self.ndcg_weighting:
            # rank within group by descending margin: stable double argsort
            order = torch.argsort(margin, descending=True, stable=True)
            perm2 = torch.argsort(group_id[order], stable=True)
            final = order[perm2]  # docs ordered by (group, margin desc)
            pos = torch.empty(n, dtype=torch.int64, device=device)
            pos[final] = torch.arange(n, device=device)
            rank = (pos - group_ptr[group_id]).to(torch.int32)
            # idcg per group from ideal (label-sorted) ordering
            lorder = torch.argsort(label, descending=True, stable=True)
            lperm2 = torch.argsort(group_id[lorder], stable=True)
            lfinal = lorder[lperm2]
            lpos = torch.empty(n, dtype=torch.int64, device=device)
            lpos[lfinal] = torch.arange(n, device=device)
            lrank = (lpos - group_ptr[group_id]).double()
            gains = torch.pow(2.0, label.double()) - 1.0
            disc = 1.0 / torch.log2(lrank + 2.0)
            idcg = torch.zeros(len(counts), dtype=torch.float64, device=device)
            idcg.scatter_add_(0, group_id, gains * disc)
        else:
            rank = torch.zeros(n, dtype=torch.int32, device=device)
            idcg = torch.zeros(1, dtype=torch.float64, device=device)

        gpair = gpu_ops.lambdarank_grad(
            margin.contiguous().float(),
            label.contiguous().float(),
            group_ptr,
            rank,
            idcg,
            self.ndcg_weighting,
        )
        if weight is not None:
            gpair = gpair * weight.unsqueeze(1)
        return gpair


class RankMAP(LambdaRankBase):
    """rank:map - pairwise lambdas weighted by |delta-AP| (binary
    relevance). Uses the generic per-group path on every device (no
    dedicated HIP kernel; MAP weighting is a rarely-hot objective)."""

    name = "rank:map"
    default_metric = "map"
    map_weighting = True

    def _gradients_gpu(self, margin, label, weight, qid):
        # no dedicated HIP kernel: run the generic group loop on the
        # device tensors (correct, just not the fast path)
        return self._gradients_generic(margin, label, weight, qid)


class RankPairwise(LambdaRankBase):
    name = "rank:pairwise"
    default_metric = "map"


class RankNDCG(LambdaRankBase):
    name = "rank:ndcg"
    ndcg_weighting = True
    default_metric = "ndcg"


class CustomObjective(Objective):
    """Wraps a user callable obj(preds, dtrain) -> (grad, hess).

    Matches the xgboost custom-objective convention the reference passes
    through to every actor (reference test_xgboost_api.py:77-152).
    """

    name = "custom"
    default_metric = "rmse"

    def __init__(self, fn, n_class=0):
        self.fn = fn
        self.n_class = n_class

    def gradients(self, margin, label, weight=None, qid=None):
        import numpy as np

        class _Shim:
            def __init__(self, label, weight):
                self._label = label
                self._weight = weight

            def get_label(self):
                return self._label

            def get_weight(self):
                return self._weight

        preds = margin.detach().cpu().numpy()
        shim = _Shim(
            label.detach().cpu().numpy(),
            None if weight is None else weight.detach().cpu().numpy(),
        )
        grad, hess = self.fn(preds, shim)
        g = torch.as_tensor(np.asarray(grad), dtype=torch.float32, device=margin.device)
        h = torch.as_tensor(np.asarray(hess), dtype=torch.float32, device=margin.device)
        return torch.stack([g.reshape(margin.shape), h.reshape(margin.shape)], dim=-1)



class Poisson(Objective):
    """count:poisson - margin is log(mean); xgboost semantics incl. the
    max_delta_step=0.7 default applied by the trainer."""

    name = "count:poisson"
    default_metric = "poisson-nloglik"

    def prob_to_margin(self, base_score):
        return math.log(max(base_score, _EPS))

    def gradients(self, margin, label, weight=None, qid=None):
        mu = elemwise_sizeinv(torch.exp, margin)
        g = mu - label
        h = mu
        return self._apply_weight(g, h, weight)

    def transform_prediction(self, margin):
        return elemwise_sizeinv(torch.exp, margin)

    def validate_label(self, label):
        if bool((label < 0).any()):
            raise ValueError("count:poisson requires non-negative labels")


class Gamma(Objective):
    """reg:gamma - log-link gamma deviance."""

    name = "reg:gamma"
    default_metric = "gamma-nloglik"

    def prob_to_margin(self, base_score):
        return math.log(max(base_score, _EPS))

    def gradients(self, margin, label, weight=None, qid=None):
        ratio = label * elemwise_sizeinv(torch.exp, -margin)
        g = 1.0 - ratio
        h = ratio
        return self._apply_weight(g, h, weight)

    def transform_prediction(self, margin):
        return elemwise_sizeinv(torch.exp, margin)

    def validate_label(self, label):
        if bool((label <= 0).any()):
            raise ValueError("reg:gamma requires positive labels")


class Tweedie(Objective):
    """reg:tweedie - log-link, variance power rho (default 1.5)."""

    name = "reg:tweedie"
    default_metric = "tweedie-nloglik"

    def __init__(self, rho: float = 1.5):
        self.rho = float(rho)
        self.default_metric = f"tweedie-nloglik@{self.rho}"

    def prob_to_margin(self, base_score):
        return math.log(max(base_score, _EPS))

    def gradients(self, margin, label, weight=None, qid=None):
        r = self.rho
        a = elemwise_sizeinv(torch.exp, (1.0 - r) * margin)
        b = elemwise_sizeinv(torch.exp, (2.0 - r) * margin)
        g = -label * a + b
        h = -label * (1.0 - r) * a + (2.0 - r) * b
        return self._apply_weight(g, h, weight)

    def transform_prediction(self, margin):
        return elemwise_sizeinv(torch.exp, margin)

    def validate_label(self, label):
        if bool((label < 0).any()):
            raise ValueError("reg:tweedie requires non-negative labels")


class SquaredLogError(Objective):
    """reg:squaredlogerror - 0.5*(log1p(pred)-log1p(y))^2 on raw margin."""

    name = "reg:squaredlogerror"
    default_metric = "rmsle"

    def gradients(self, margin, label, weight=None, qid=None):
        p1 = torch.clamp(margin + 1.0, min=1e-6)
        d = elemwise_sizeinv(torch.log, p1) - elemwise_sizeinv(
            torch.log, torch.clamp(label + 1.0, min=1e-6)
        )
        g = d / p1
        h = torch.clamp((1.0 - d) / (p1 * p1), min=1e-6)
        return self._apply_weight(g, h, weight)

    def validate_label(self, label):
        if bool((label < -1).any()):
            raise ValueError("reg:squaredlogerror requires label > -1")



class AFT(Objective):
    """survival:aft - accelerated failure time with interval censoring.

    margin predicts log(survival time); labels arrive as
    (label_lower_bound, label_upper_bound) columns (reference passes them
    through RayDMatrix, matrix.py label-bounds fields). Uncensored rows
    have lower == upper; right-censored have upper == +inf; left-censored
    lower <= 0. Distributions: normal / logistic / extreme (Gumbel),
    scale sigma (xgboost aft_loss_distribution[_scale] params).
    The engine passes label as an [n, 2] stack of the bounds.
    """

    name = "survival:aft"
    default_metric = "aft-nloglik"

    def __init__(self, dist: str = "normal", sigma: float = 1.0):
        self.dist = dist
        self.sigma = float(sigma)

    def prob_to_margin(self, base_score):
        return math.log(max(base_score, _EPS))

    def _logpdf_terms(self, z):
        """Returns (f(z), dlogf(z)) for the chosen distribution."""
        if self.dist == "logistic":
            s = sigmoid_sizeinv(z)
            f = s * (1.0 - s)
            dlogf = 1.0 - 2.0 * s
        elif self.dist == "extreme":
            ez = elemwise_sizeinv(torch.exp, torch.clamp(z, max=15.0))
            f = ez * elemwise_sizeinv(torch.exp, -ez)
            dlogf = 1.0 - ez
        else:  # normal
            f = elemwise_sizeinv(torch.exp, -0.5 * z * z) / math.sqrt(
                2.0 * math.pi
            )
            dlogf = -z
        return f, dlogf

    def _cdf(self, z):
        if self.dist == "logistic":
            return sigmoid_sizeinv(z)
        if self.dist == "extreme":
            ez = elemwise_sizeinv(torch.exp, torch.clamp(z, max=15.0))
            return 1.0 - elemwise_sizeinv(torch.exp, -ez)
        return elemwise_sizeinv(torch.special.ndtr, z)

    def nll(self, margin, yl, yu):
        """Per-row negative log likelihood (metric + gradients share it)."""
        sig = self.sigma
        m = margin
        lo = torch.log(torch.clamp(yl, min=_EPS))
        hi = torch.log(torch.clamp(yu, min=_EPS))
        zl = torch.clamp((lo - m) / sig, min=-15.0, max=15.0)
        zu = torch.clamp((hi - m) / sig, min=-15.0, max=15.0)
        uncensored = yl == yu
        right = torch.isinf(yu)
        left = yl <= 0
        f_l, _ = self._logpdf_terms(zl)
        F_l = self._cdf(zl)
        F_u = self._cdf(zu)
        D = torch.clamp(
            torch.where(right, 1.0 - F_l,
                        torch.where(left, F_u, F_u - F_l)),
            min=1e-12,
        )
        nll_cens = -torch.log(D)
        nll_unc = -torch.log(torch.clamp(f_l, min=1e-12)) + math.log(
            sig
        ) + lo
        return torch.where(uncensored, nll_unc, nll_cens)

    def _survival(self, z):
        """S(z) = 1 - F(z), computed without cancellation."""
        if self.dist == "logistic":
            return sigmoid_sizeinv(-z)
        if self.dist == "extreme":
            ez = elemwise_sizeinv(torch.exp, torch.clamp(z, max=15.0))
            return elemwise_sizeinv(torch.exp, -ez)
        return elemwise_sizeinv(torch.special.ndtr, -z)

    def _hazard(self, z):
        """f(z)/S(z) and its z-derivative, in stable closed forms."""
        if self.dist == "logistic":
            hz = sigmoid_sizeinv(z)
            dhz = hz * (1.0 - hz)
        elif self.dist == "extreme":
            hz = elemwise_sizeinv(torch.exp, torch.clamp(z, max=15.0))
            dhz = hz
        else:  # normal: hz' = hz*(hz - z)
            f, _ = self._logpdf_terms(z)
            hz = f / torch.clamp(self._survival(z), min=1e-300)
            dhz = hz * (hz - z)
        return hz, dhz

    def _rev_hazard(self, z):
        """f(z)/F(z) and its z-derivative (left censoring)."""
        if self.dist == "logistic":
            rh = sigmoid_sizeinv(-z)
            drh = -rh * (1.0 - rh)
        elif self.dist == "extreme":
            # f/F = e^z e^{-e^z} / (1 - e^{-e^z})
            f, _ = self._logpdf_terms(z)
            F = torch.clamp(1.0 - self._survival(z), min=1e-300)
            rh = f / F
            ez = elemwise_sizeinv(torch.exp, torch.clamp(z, max=15.0))
            drh = rh * (1.0 - ez - rh)
        else:  # normal: rh' = rh*(-z - rh)
            f, _ = self._logpdf_terms(z)
            F = torch.clamp(
                elemwise_sizeinv(torch.special.ndtr, z), min=1e-300
            )
            rh = f / F
            drh = rh * (-z - rh)
        return rh, drh

    def gradients(self, margin, label, weight=None, qid=None):
        # label is [n, 2]: (lower, upper); all math in f64 - the censored
        # branches divide underflow-prone densities
        yl, yu = label[:, 0].double(), label[:, 1].double()
        m = margin.double()
        sig = self.sigma
        lo = torch.log(torch.clamp(yl, min=_EPS))
        hi = torch.log(torch.clamp(yu, min=_EPS))
        zl = torch.clamp((lo - m) / sig, min=-15.0, max=15.0)
        zu = torch.clamp((hi - m) / sig, min=-15.0, max=15.0)
        uncensored = yl == yu
        right = torch.isinf(yu)
        left = yl <= 0
        interval = ~(uncensored | right | left)

        # uncensored: nll = -log f(z)
        f_l, dlogf_l = self._logpdf_terms(zl)
        g_unc = dlogf_l / sig
        if self.dist == "logistic":
            su = sigmoid_sizeinv(zl)
            h_unc = 2.0 * su * (1.0 - su) / (sig * sig)
        elif self.dist == "extreme":
            h_unc = elemwise_sizeinv(
                torch.exp, torch.clamp(zl, max=15.0)
            ) / (sig * sig)
        else:
            h_unc = torch.full_like(zl, 1.0 / (sig * sig))

        # right-censored: nll = -log S(zl); g = -hazard/sig
        hz, dhz = self._hazard(zl)
        g_right = -hz / sig
        h_right = dhz / (sig * sig)

        # left-censored: nll = -log F(zu); g = +rev_hazard/sig
        rh, drh = self._rev_hazard(zu)
        g_left = rh / sig
        h_left = -drh / (sig * sig)

        # interval: nll = -log(F(zu) - F(zl))
        f_u, dlogf_u = self._logpdf_terms(zu)
        D = torch.clamp(
            self._survival(zl) - self._survival(zu), min=1e-300
        )
        g_int = (f_u - f_l) / (sig * D)
        fpu = f_u * dlogf_u
        fpl = f_l * dlogf_l
        h_int = ((f_u - f_l) ** 2 / (D * D) - (fpu - fpl) / D) / (
            sig * sig
        )

        g = torch.where(
            uncensored, g_unc,
            torch.where(right, g_right,
                        torch.where(left, g_left, g_int)),
        ).float()
        h = torch.where(
            uncensored, h_unc,
            torch.where(right, h_right,
                        torch.where(left, h_left, h_int)),
        )
        h = torch.clamp(h, min=1e-6).float()
        return self._apply_weight(g, h, weight)

    def transform_prediction(self, margin):
        return elemwise_sizeinv(torch.exp, margin)

    def validate_label(self, label):
        pass  # bounds validated at stack time



class Cox(Objective):
    """survival:cox - Breslow partial likelihood on right-censored data.

    xgboost label encoding: label > 0 is an event at time t=label;
    label < 0 is right-censored at t=|label|. Risk sets are computed over
    the local shard (xgboost's distributed behavior); margins are
    log-hazard ratios, predictions exp(margin). All ops are sorted
    cumsums - deterministic on CPU and GPU, no custom kernels.
    """

    name = "survival:cox"
    default_metric = "cox-nloglik"

    def gradients(self, margin, label, weight=None, qid=None):
        t = label.abs().double()
        event = (label > 0).double()
        m = margin.double()
        eta = elemwise_sizeinv(torch.exp, m - m.max())
        # sort by time DESCENDING: risk set of time t = all rows with
        # t_j >= t = prefix of the sorted order
        order = torch.argsort(-t, stable=True)
        t_s = t[order]
        eta_s = eta[order]
        ev_s = event[order]
        cum_eta = torch.cumsum(eta_s, 0)
        # rows tied on time share one risk set: take the cumsum at the
        # LAST row of each tied block
        change = torch.ones_like(t_s, dtype=torch.bool)
        change[:-1] = t_s[:-1] != t_s[1:]
        block_last = torch.cumsum(change.long(), 0)  # 1-based block id
        # last index of each block via scatter of positions
        n = t_s.numel()
        pos = torch.arange(n, device=t.device)
        nblocks = int(block_last[-1])
        last_idx = torch.zeros(
            nblocks + 1, dtype=torch.long, device=t.device
        )
        last_idx.scatter_reduce_(
            0, block_last, pos, reduce="amax", include_self=False
        )
        R_s = cum_eta[last_idx[block_last]]
        R_s = torch.clamp(R_s, min=1e-300)
        # event terms: per sorted row, a_k = ev/R, b_k = ev/R^2;
        # each row i accumulates over events with t_k <= t_i, i.e. the
        # SUFFIX of the descending order
        a = ev_s / R_s
        b = ev_s / (R_s * R_s)
        # suffix sums respecting ties: all rows of a tied block get the
        # same suffix (events at equal time include each other's risk)
        suf_a = torch.flip(torch.cumsum(torch.flip(a, [0]), 0), [0])
        suf_b = torch.flip(torch.cumsum(torch.flip(b, [0]), 0), [0])
        # align tied blocks to their FIRST row's suffix value
        first_idx = torch.zeros(
            nblocks + 1, dtype=torch.long, device=t.device
        )
        first_idx.scatter_reduce_(
            0, block_last, pos, reduce="amin", include_self=False
        )
        suf_a = suf_a[first_idx[block_last]]
        suf_b = suf_b[first_idx[block_last]]
        g_s = -ev_s + eta_s * suf_a
        h_s = torch.clamp(
            eta_s * suf_a - eta_s * eta_s * suf_b, min=1e-6
        )
        g = torch.empty_like(g_s)
        h = torch.empty_like(h_s)
        g[order] = g_s
        h[order] = h_s
        return self._apply_weight(g.float(), h.float(), weight)

    def transform_prediction(self, margin):
        return elemwise_sizeinv(torch.exp, margin)

    def validate_label(self, label):
        if bool((label == 0).any()):
            raise ValueError(
                "survival:cox labels must be nonzero (+t event, -t "
                "censored)"
            )



class PseudoHuber(Objective):
    """reg:pseudohubererror - smooth Huber, slope delta (huber_slope)."""

    name = "reg:pseudohubererror"
    default_metric = "mphe"

    def __init__(self, delta: float = 1.0):
        self.delta = float(delta)

    def gradients(self, margin, label, weight=None, qid=None):
        r = margin - label
        z = 1.0 + (r / self.delta) ** 2
        sq = torch.sqrt(z)
        g = r / sq
        h = torch.clamp(1.0 / (z * sq), min=_EPS)
        return self._apply_weight(g, h, weight)


class Hinge(Objective):
    """binary:hinge - predicts 0/1 directly."""

    name = "binary:hinge"
    default_metric = "error"

    def gradients(self, margin, label, weight=None, qid=None):
        y = 2.0 * label - 1.0  # {0,1} -> {-1,+1}
        active = margin * y < 1.0
        g = torch.where(active, -y, torch.zeros_like(margin))
        h = torch.where(
            active, torch.ones_like(margin),
            torch.full_like(margin, _EPS),
        )
        return self._apply_weight(g, h, weight)

    def transform_prediction(self, margin):
        return (margin > 0).to(torch.float32)

    def validate_label(self, label):
        if bool((label < 0).any()) or bool((label > 1).any()):
            raise ValueError("binary:hinge labels must be 0/1")


_REGISTRY = {
    "reg:squarederror": SquaredError,
    "reg:linear": SquaredError,  # legacy alias
    "reg:absoluteerror": AbsoluteError,
    "reg:logistic": RegLogistic,
    "binary:logistic": Logistic,
    "binary:logitraw": LogisticRaw,
    "rank:pairwise": RankPairwise,
    "rank:ndcg": RankNDCG,
    "rank:map": RankMAP,
    "count:poisson": Poisson,
    "reg:gamma": Gamma,
    "reg:tweedie": Tweedie,
    "reg:squaredlogerror": SquaredLogError,
    "survival:aft": AFT,
    "survival:cox": Cox,
    "reg:pseudohubererror": PseudoHuber,
    "binary:hinge": Hinge,
}


def get_objective(
    name_or_fn, num_class: int = 0, scale_pos_weight: float = 1.0,
    tweedie_variance_power: float = 1.5,
) -> Objective:
    if callable(name_or_fn):
        return CustomObjective(name_or_fn, num_class)
    name = name_or_fn or "reg:squarederror"
    if name == "custom":
        # a model TRAINED with a custom python objective: predictions are
        # raw margins (xgboost semantics - no PredTransform is known), and
        # base_score passes through untransformed.
        o = Objective()
        o.name = "custom"
        o.n_class = num_class
        return o
    if name == "reg:tweedie":
        return Tweedie(tweedie_variance_power)
    if name == "survival:aft":
        return AFT()  # dist/sigma set by the engine from params
    if name in ("multi:softmax",):
        return SoftmaxClass(num_class)
    if name in ("multi:softprob",):
        return SoftProb(num_class)
    if name not in _REGISTRY:
        raise ValueError(f"Unsupported objective: {name}")
    cls = _REGISTRY[name]
    if issubclass(cls, Logistic) and scale_pos_weight != 1.0:
        return cls(scale_pos_weight)
    return cls()
