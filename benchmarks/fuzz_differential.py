"""Round-2 randomized differential sweep v2 (correct invariants):
- no-subsample configs: world-1 == world-2 bitwise
- subsample configs: per-world determinism (run twice, bitwise)
- ubj round-trip bitwise; resume bitwise (gbtree) / 1-ulp (dart)
"""
import random
import sys

import numpy as np

import os as _os
_sys_root = _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__)))
sys.path.insert(0, _sys_root)
from tests.utils import create_data
from xgboost_ray_amd import RayDMatrix, RayParams, train
from xgboost_ray_amd.booster import Booster


def main():
    import argparse

    ap = argparse.ArgumentParser()
    ap.add_argument("--seed", type=int, default=7)
    ap.add_argument("--trials", type=int, default=24)
    args = ap.parse_args()
    random.seed(args.seed)
    fails = []
    for trial in range(args.trials):
        cfg = {"objective": "binary:logistic",
               "max_depth": random.choice([3, 5, 7]),
               "eta": random.choice([0.1, 0.3]), "seed": trial}
        if random.random() < 0.35:
            cfg["subsample"] = random.choice([0.6, 0.8])
        if random.random() < 0.4:
            cfg["colsample_bytree"] = 0.7
        if random.random() < 0.3:
            cfg["colsample_bynode"] = 0.6
        if random.random() < 0.3:
            cfg["booster"] = "dart"
            cfg["rate_drop"] = 0.3
            if random.random() < 0.5:
                cfg["sample_type"] = "weighted"
        if random.random() < 0.25:
            cfg["num_parallel_tree"] = 2
        if random.random() < 0.25:
            cfg["reg_alpha"] = 0.3
            cfg["gamma"] = 0.2
        if random.random() < 0.2:
            cfg["monotone_constraints"] = "(1,0,-1,0,0,0)"
        rounds = random.choice([4, 7])
        X, y = create_data(random.choice([3000, 6000]), 6, seed=trial)

        def tr(actors, r=rounds, model=None):
            return train(dict(cfg), RayDMatrix(X, label=y), r,
                         ray_params=RayParams(num_actors=actors),
                         xgb_model=model)

        try:
            b2 = tr(2)
            p2 = b2.predict(X, output_margin=True)
            if "subsample" in cfg:
                rep = tr(2).predict(X, output_margin=True)
                ok = np.array_equal(p2, rep)
                tag = "per-world determinism"
            else:
                p1 = tr(1).predict(X, output_margin=True)
                ok = np.array_equal(p1, p2)
                tag = "world invariance"
            if not ok:
                fails.append((trial, cfg, tag))
                continue
            rt = Booster().load_model(b2.save_raw("ubj"))
            if not np.array_equal(p2, rt.predict(X, output_margin=True)):
                fails.append((trial, cfg, "ubj roundtrip"))
                continue
            half = tr(2, rounds // 2)
            res = tr(2, rounds - rounds // 2, model=half)
            pr = res.predict(X, output_margin=True)
            if cfg.get("booster") == "dart":
                ok = np.allclose(p2, pr, atol=2e-6, rtol=1e-6)
            else:
                ok = np.array_equal(p2, pr)
            if not ok:
                fails.append((trial, cfg, "resume"))
        except Exception as e:
            fails.append((trial, cfg, repr(e)[:120]))
    print("FUZZ RESULT (seed %d, %d trials):" % (args.seed, args.trials),
          "ALL PASS" if not fails else fails)


if __name__ == "__main__":
    main()
