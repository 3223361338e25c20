"""Evaluation metrics, designed for distributed evaluation.

Each metric reduces its local shard to a small float64 stats vector that is
summed across workers with one AllReduce (reference semantics: XGBoost's
per-round metric AllReduce inside Rabit), then
finalized identically on every rank. AUC is computed from a fixed
score-histogram (16384 bins) so it needs no global sort; mAP/NDCG reduce
per-query sums.
"""

from typing import Optional

import torch

from xgboost_ray_amd.engine.objectives import sigmoid_sizeinv

_AUC_BINS = 16384


def _fused_eval_ok(margin):
    """GPU fused metric kernels: one pass instead of 4-6 f64 torch
    kernels over the shard (RXGB_FUSED_EVAL=0 disables)."""
    import os

    return (
        margin.is_cuda
        and margin.dim() == 1
        and os.environ.get("RXGB_FUSED_EVAL") != "0"
    )


class Metric:
    name = "base"
    higher_better = False

    def local_stats(self, margin, label, weight, qid, obj) -> torch.Tensor:
        raise NotImplementedError

    def finalize(self, stats: torch.Tensor) -> float:
        raise NotImplementedError


def _w(label, weight):
    if weight is None:
        return torch.ones_like(label, dtype=torch.float64)
    return weight.double()


def _eval_pred(margin, obj):
    """xgboost's EvalTransform semantics: metrics defined on predictions
    (rmse/mae/mphe) receive the objective's transformed output — sigmoid for
    logistic, exp for log-link objectives — before the metric is computed
    (XGBoost learner.cc EvalOneIter -> obj_->EvalTransform). Falls back to
    the raw margin when no objective is known or the transform changes the
    shape (multiclass prob matrices, custom objectives)."""
    p = margin.double()
    if obj is None:
        return p
    try:
        t = obj.transform_prediction(p)
    except Exception:
        return p
    return t.double() if t.shape == p.shape else p


class RMSE(Metric):
    name = "rmse"

    def local_stats(self, margin, label, weight, qid, obj):
        pred = _eval_pred(margin, obj)
        w = _w(label, weight)
        se = (w * (pred - label.double()) ** 2).sum()
        return torch.stack([se, w.sum()])

    def finalize(self, s):
        return float(torch.sqrt(s[0] / s[1]))


class MAE(Metric):
    name = "mae"

    def local_stats(self, margin, label, weight, qid, obj):
        w = _w(label, weight)
        pred = _eval_pred(margin, obj)
        return torch.stack(
            [(w * (pred - label.double()).abs()).sum(), w.sum()]
        )

    def finalize(self, s):
        return float(s[0] / s[1])


class LogLoss(Metric):
    name = "logloss"

    def local_stats(self, margin, label, weight, qid, obj):
        if _fused_eval_ok(margin):
            from xgboost_ray_amd.ops import gpu

            return gpu.eval_logloss(margin, label, weight)
        p = sigmoid_sizeinv(margin.double()).clamp(1e-16, 1 - 1e-16)
        y = label.double()
        w = _w(label, weight)
        ll = -(y * torch.log(p) + (1 - y) * torch.log(1 - p))
        return torch.stack([(w * ll).sum(), w.sum()])

    def finalize(self, s):
        return float(s[0] / s[1])


class BinaryError(Metric):
    name = "error"

    def __init__(self, threshold: float = 0.5):
        self.threshold = threshold
        if threshold != 0.5:
            self.name = f"error@{threshold}"

    def local_stats(self, margin, label, weight, qid, obj):
        p = sigmoid_sizeinv(margin.double())
        w = _w(label, weight)
        wrong = (p > self.threshold).double() != label.double()
        return torch.stack([(w * wrong).sum(), w.sum()])

    def finalize(self, s):
        return float(s[0] / s[1])


class AUC(Metric):
    """Binned ROC AUC: weighted pos/neg histograms over sigmoid(margin)."""

    name = "auc"
    higher_better = True

    def local_stats(self, margin, label, weight, qid, obj):
        if weight is None and _fused_eval_ok(margin):
            from xgboost_ray_amd.ops import gpu

            hist = gpu.eval_auc_hist(margin, label, _AUC_BINS)
            # kernel layout [neg | pos] -> metric layout [pos | neg]
            return torch.cat(
                [hist[_AUC_BINS:], hist[:_AUC_BINS]]
            ).double()
        p = sigmoid_sizeinv(margin.double())
        b = torch.clamp((p * _AUC_BINS).long(), max=_AUC_BINS - 1)
        if weight is None:
            # unweighted: integer scatter (native u64 atomics on GPU -
            # the f64 scatter goes through CAS loops and is ~50x slower).
            # One fused scatter into [neg plane | pos plane]: boolean
            # fancy-indexing (b[ypos]) costs a nonzero + sort + two
            # gathers per eval round.
            idx = b + (label > 0.5).long() * _AUC_BINS
            hist = torch.zeros(
                2 * _AUC_BINS, dtype=torch.int64, device=margin.device
            )
            src = torch.ones(
                (), dtype=torch.int64, device=margin.device
            ).expand(idx.shape)
            hist.scatter_add_(0, idx, src)
            return torch.cat([hist[_AUC_BINS:], hist[:_AUC_BINS]]).double()
        w = _w(label, weight)
        pos = torch.zeros(_AUC_BINS, dtype=torch.float64, device=margin.device)
        neg = torch.zeros(_AUC_BINS, dtype=torch.float64, device=margin.device)
        y = label.double()
        pos.scatter_add_(0, b, w * y)
        neg.scatter_add_(0, b, w * (1 - y))
        return torch.cat([pos, neg])

    def finalize(self, s):
        pos, neg = s[:_AUC_BINS], s[_AUC_BINS:]
        # iterate bins ascending: AUC = sum over bins of
        # neg_below * pos_here + 0.5 * neg_here * pos_here
        cneg = torch.cumsum(neg, 0) - neg
        auc = (cneg * pos).sum() + 0.5 * (neg * pos).sum()
        denom = pos.sum() * neg.sum()
        return float(auc / denom) if denom > 0 else 0.5


class AUCPR(Metric):
    """Binned PR AUC: precision-recall area via the same score bins as AUC
    (xgboost `aucpr`; reference passes eval_metric through, SURVEY #2.3)."""

    name = "aucpr"
    higher_better = True

    def local_stats(self, margin, label, weight, qid, obj):
        return AUC.local_stats(self, margin, label, weight, qid, obj)

    def finalize(self, s):
        pos, neg = s[:_AUC_BINS], s[_AUC_BINS:]
        tot_pos = pos.sum()
        if float(tot_pos) <= 0:
            return 0.0
        # descending-score sweep: cumulative TP/FP from the top bin down
        tp = torch.flip(torch.cumsum(torch.flip(pos, [0]), 0), [0])
        fp = torch.flip(torch.cumsum(torch.flip(neg, [0]), 0), [0])
        recall = tp / tot_pos
        prec = tp / torch.clamp(tp + fp, min=1e-300)
        # integrate precision d(recall) over bins, descending threshold:
        # recall increases as the threshold drops
        r = torch.flip(recall, [0])
        p = torch.flip(prec, [0])
        dr = torch.diff(r, prepend=torch.zeros(1, dtype=r.dtype,
                                               device=r.device))
        return float((p * dr).sum())


class MLogLoss(Metric):
    name = "mlogloss"

    def local_stats(self, margin, label, weight, qid, obj):
        logp = torch.log_softmax(margin.double(), dim=1)
        w = _w(label, weight)
        ll = -logp[torch.arange(len(label), device=margin.device), label.long()]
        return torch.stack([(w * ll).sum(), w.sum()])

    def finalize(self, s):
        return float(s[0] / s[1])


class MError(Metric):
    name = "merror"

    def local_stats(self, margin, label, weight, qid, obj):
        pred = margin.argmax(dim=1)
        w = _w(label, weight)
        wrong = (pred != label.long()).double()
        return torch.stack([(w * wrong).sum(), w.sum()])

    def finalize(self, s):
        return float(s[0] / s[1])


class _PerGroupMetric(Metric):
    higher_better = True

    def group_score(self, m, y) -> float:
        raise NotImplementedError

    def local_stats(self, margin, label, weight, qid, obj):
        if qid is None:
            qid = torch.zeros(len(label), dtype=torch.int64, device=margin.device)
        total = torch.zeros(2, dtype=torch.float64, device=margin.device)
        change = torch.ones_like(qid, dtype=torch.bool)
        change[1:] = qid[1:] != qid[:-1]
        starts = torch.nonzero(change).flatten().tolist() + [len(qid)]
        for gi in range(len(starts) - 1):
            s, e = starts[gi], starts[gi + 1]
            total[0] += self.group_score(margin[s:e], label[s:e])
            total[1] += 1
        return total

    def finalize(self, s):
        return float(s[0] / s[1]) if s[1] > 0 else 0.0


class NDCG(_PerGroupMetric):
    name = "ndcg"

    def __init__(self, k: Optional[int] = None):
        self.k = k
        if k:
            self.name = f"ndcg@{k}"

    def group_score(self, m, y):
        n = len(y)
        k = min(self.k or n, n)
        order = torch.argsort(m.double(), descending=True, stable=True)
        gains = torch.pow(2.0, y.double()) - 1.0
        disc = 1.0 / torch.log2(
            torch.arange(n, device=m.device, dtype=torch.float64) + 2.0
        )
        dcg = (gains[order][:k] * disc[:k]).sum()
        ideal, _ = torch.sort(gains, descending=True)
        idcg = (ideal[:k] * disc[:k]).sum()
        return float(dcg / idcg) if idcg > 0 else 1.0


class MAP(_PerGroupMetric):
    name = "map"

    def group_score(self, m, y):
        order = torch.argsort(m.double(), descending=True, stable=True)
        rel = (y[order] > 0).double()
        if rel.sum() == 0:
            return 1.0
        csum = torch.cumsum(rel, 0)
        ranks = torch.arange(1, len(y) + 1, device=m.device, dtype=torch.float64)
        ap = ((csum / ranks) * rel).sum() / rel.sum()
        return float(ap)



class PoissonNLogLik(Metric):
    name = "poisson-nloglik"
    higher_better = False

    def local_stats(self, margin, label, weight, qid, obj):
        mu = torch.exp(margin.double()).clamp(min=1e-16)
        y = label.double()
        nll = mu - y * torch.log(mu) + torch.lgamma(y + 1.0)
        w = _w(label, weight)
        return torch.stack([(w * nll).sum(), w.sum()])

    def finalize(self, s):
        return float(s[0] / s[1])


class GammaNLogLik(Metric):
    name = "gamma-nloglik"
    higher_better = False

    def local_stats(self, margin, label, weight, qid, obj):
        mu = torch.exp(margin.double()).clamp(min=1e-16)
        y = label.double()
        nll = y / mu + torch.log(mu)
        w = _w(label, weight)
        return torch.stack([(w * nll).sum(), w.sum()])

    def finalize(self, s):
        return float(s[0] / s[1])


class TweedieNLogLik(Metric):
    higher_better = False

    def __init__(self, rho: float = 1.5):
        self.rho = float(rho)
        self.name = f"tweedie-nloglik@{self.rho}"

    def local_stats(self, margin, label, weight, qid, obj):
        m = margin.double()
        y = label.double()
        r = self.rho
        nll = -y * torch.exp((1.0 - r) * m) / (1.0 - r) + torch.exp(
            (2.0 - r) * m
        ) / (2.0 - r)
        w = _w(label, weight)
        return torch.stack([(w * nll).sum(), w.sum()])

    def finalize(self, s):
        return float(s[0] / s[1])


class RMSLE(Metric):
    name = "rmsle"
    higher_better = False

    def local_stats(self, margin, label, weight, qid, obj):
        p = torch.clamp(_eval_pred(margin, obj), min=-1.0 + 1e-6)
        d = torch.log1p(p) - torch.log1p(label.double())
        w = _w(label, weight)
        return torch.stack([(w * d * d).sum(), w.sum()])

    def finalize(self, s):
        import math

        return float(math.sqrt(s[0] / s[1]))



class AFTNLogLik(Metric):
    """survival:aft negative log likelihood; label is the [n, 2]
    (lower, upper) bound stack the engine builds for AFT training."""

    name = "aft-nloglik"
    higher_better = False

    def local_stats(self, margin, label, weight, qid, obj):
        from xgboost_ray_amd.engine.objectives import AFT

        o = obj if isinstance(obj, AFT) else AFT()
        nll = o.nll(margin.double(), label[:, 0].double(),
                    label[:, 1].double())
        w = _w(label[:, 0], weight)
        return torch.stack([(w * nll).sum(), w.sum()])

    def finalize(self, s):
        return float(s[0] / s[1])



class CoxNLogLik(Metric):
    """Breslow negative partial log likelihood per event (local shard)."""

    name = "cox-nloglik"
    higher_better = False

    def local_stats(self, margin, label, weight, qid, obj):
        t = label.abs().double()
        event = label > 0
        m = margin.double()
        shift = m.max()
        eta = torch.exp(m - shift)
        order = torch.argsort(-t, stable=True)
        eta_s = eta[order]
        ev_s = event[order]
        m_s = m[order]
        cum = torch.cumsum(eta_s, 0)
        ll = (m_s[ev_s] - shift - torch.log(
            torch.clamp(cum[ev_s], min=1e-300)
        )).sum()
        n_ev = ev_s.sum()
        return torch.stack([-ll, n_ev.double()])

    def finalize(self, s):
        return float(s[0] / max(float(s[1]), 1.0))



class MAPE(Metric):
    """Mean absolute percentage error (xgboost `mape`)."""

    name = "mape"
    higher_better = False

    def local_stats(self, margin, label, weight, qid, obj):
        pred = _eval_pred(margin, obj)
        y = label.double()
        w = _w(label, weight)
        ape = (pred - y).abs() / y.abs().clamp(min=1e-10)
        return torch.stack([(w * ape).sum(), w.sum()])

    def finalize(self, s):
        return float(s[0] / s[1])


class MPHE(Metric):
    """Mean pseudo-Huber error (delta = 1)."""

    name = "mphe"
    higher_better = False

    def local_stats(self, margin, label, weight, qid, obj):
        r = (margin - label).double()
        loss = torch.sqrt(1.0 + r * r) - 1.0
        w = _w(label, weight)
        return torch.stack([(w * loss).sum(), w.sum()])

    def finalize(self, s):
        return float(s[0] / s[1])


def get_metric(name: str) -> Metric:
    if name.startswith("error@"):
        return BinaryError(float(name.split("@")[1]))
    if name.startswith("ndcg@"):
        return NDCG(int(name.split("@")[1]))
    if name.startswith("tweedie-nloglik"):
        parts = name.split("@")
        return TweedieNLogLik(float(parts[1]) if len(parts) > 1 else 1.5)
    table = {
        "rmse": RMSE,
        "mae": MAE,
        "logloss": LogLoss,
        "error": BinaryError,
        "auc": AUC,
        "aucpr": AUCPR,
        "mlogloss": MLogLoss,
        "merror": MError,
        "ndcg": NDCG,
        "map": MAP,
        "poisson-nloglik": PoissonNLogLik,
        "gamma-nloglik": GammaNLogLik,
        "rmsle": RMSLE,
        "aft-nloglik": AFTNLogLik,
        "cox-nloglik": CoxNLogLik,
        "mphe": MPHE,
        "mape": MAPE,
    }
    if name not in table:
        raise ValueError(f"Unsupported eval_metric: {name}")
    return table[name]()
