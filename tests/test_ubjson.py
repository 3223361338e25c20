"""UBJSON model format: codec correctness, save/load matrix across
{json, ubj, xgb} x {gbtree, dart, gblinear, multiclass}, and a vendored
golden XGBoost-schema model that must survive load/save unchanged.

The format is the reference's user contract: ``bst.save_model("model.xgb")``
(reference README.md:78) and Tune's ``load_model`` (reference
xgboost_ray/tune.py:130-156) feed stock XGBoost's strict per-extension
parsers, so ``.ubj``/``.xgb`` must hold real UBJSON, not JSON bytes.
"""

import json
import os
import struct

import numpy as np
import pytest
import torch

from tests.utils import create_data
from xgboost_ray_amd import ubjson
from xgboost_ray_amd.booster import Booster
from xgboost_ray_amd.engine.quantile import BinnedMatrix
from xgboost_ray_amd.engine.trainer import run_training

GOLDEN = os.path.join(os.path.dirname(__file__), "data",
                      "golden_xgboost_model.json")


class TestCodec:
    def test_primitives_roundtrip(self):
        doc = {
            "s": "héllo ünïcode",
            "i_small": 7, "i_neg": -100, "u8": 200, "i16": 30000,
            "i32": 2**30, "i64": 2**40, "huge": 2**70,
            "f": 0.1, "t": True, "f2": False, "z": None,
            "nested": {"a": [1, "two", 3.0, None, {"b": []}]},
            "empty": {},
        }
        assert ubjson.loads(ubjson.dumps(doc)) == doc

    def test_typed_arrays_roundtrip_exact(self):
        ints = [0, -5, 2**31 - 1, -(2**31)]
        big = [2**40, -(2**40)]
        floats = [0.1, -1e300, 1e-300, 3.5]
        doc = {"i32": ints, "i64": big, "f64": floats}
        out = ubjson.loads(ubjson.dumps(doc))
        assert out["i32"] == ints
        assert out["i64"] == big
        assert out["f64"] == floats  # f64 typed array: bit-exact

    def test_numpy_arrays(self):
        doc = {
            "f32": np.array([1.5, -2.25, np.nan], np.float32),
            "f64": np.array([0.1, 1e300], np.float64),
            "i32": np.array([1, -7], np.int32),
            "i64": np.array([2**40], np.int64),
            "u8": np.array([0, 255], np.uint8),
        }
        out = ubjson.loads(ubjson.dumps(doc))
        assert out["f32"][0] == 1.5 and np.isnan(out["f32"][2])
        assert out["f64"] == [0.1, 1e300]
        assert out["i32"] == [1, -7]
        assert out["i64"] == [2**40]
        assert out["u8"] == [0, 255]

    def test_known_bytes(self):
        """Byte-level spec check, independent of the decoder."""
        assert ubjson.dumps({"a": 1}) == b"{i\x01ai\x01}"
        assert ubjson.dumps("hi") == b"Si\x02hi"
        assert ubjson.dumps(True) == b"T"
        assert ubjson.dumps(None) == b"Z"
        assert ubjson.dumps(3.5) == b"D" + struct.pack("<d", 3.5)
        assert (ubjson.dumps([1.0, 2.0])
                == b"[$D#i\x02" + struct.pack("<dd", 1.0, 2.0))

    def test_decodes_foreign_markers(self):
        """Forms our encoder never emits but the spec (and other writers,
        e.g. stock XGBoost) may produce: f32 scalars, int16 lengths,
        sized-only containers, no-ops, char, high-precision."""
        # f32 scalar
        assert ubjson.loads(b"d" + struct.pack("<f", 2.5)) == 2.5
        # string with int16 length marker
        assert ubjson.loads(b"SI" + struct.pack("<h", 3) + b"abc") == "abc"
        # sized-only array: [#i2 i1 i2
        assert ubjson.loads(b"[#i\x02i\x01i\x02") == [1, 2]
        # sized+typed object: {$i#i2 <key>1<key>2
        assert ubjson.loads(
            b"{$i#i\x02i\x01a\x01i\x01b\x02") == {"a": 1, "b": 2}
        # no-ops between values
        assert ubjson.loads(b"[Ni\x01NNi\x02N]") == [1, 2]
        # char + high-precision number
        assert ubjson.loads(b"Ca") == "a"
        assert ubjson.loads(b"Hi\x0212") == 12
        # f32 typed array
        raw = b"[$d#i\x02" + struct.pack("<ff", 1.5, -0.5)
        assert ubjson.loads(raw) == [1.5, -0.5]

    def test_truncated_raises(self):
        good = ubjson.dumps({"a": [1.0, 2.0, 3.0]})
        with pytest.raises(ValueError):
            ubjson.loads(good[:-3])


def _train(kind="binary", rounds=5, **params):
    X, y = create_data(800, 5, kind=kind)
    dm = BinnedMatrix.build(
        torch.from_numpy(X), label=torch.from_numpy(y), max_bin=64,
        keep_raw=params.get("booster") == "gblinear",
    )
    base = {"objective": "binary:logistic", "max_depth": 4, "eta": 0.3}
    base.update(params)
    return run_training(base, dm, rounds), X


MODEL_CASES = [
    ("gbtree", {}),
    ("dart", {"booster": "dart", "rate_drop": 0.3}),
    ("gblinear", {"booster": "gblinear"}),
    ("multiclass", {"objective": "multi:softprob", "num_class": 4}),
    ("parallel_tree", {"num_parallel_tree": 3}),
]


class TestModelFormats:
    @pytest.mark.parametrize("ext", ["json", "ubj", "xgb"])
    @pytest.mark.parametrize("name,params", MODEL_CASES)
    def test_save_load_matrix(self, tmp_path, ext, name, params):
        kind = "multi" if name == "multiclass" else "binary"
        bst, X = _train(kind=kind, **params)
        path = str(tmp_path / f"m_{name}.{ext}")
        bst.save_model(path)
        bst2 = Booster().load_model(path)
        np.testing.assert_array_equal(
            bst.predict(X, output_margin=True),
            bst2.predict(X, output_margin=True),
        )
        # second generation must be byte-stable
        p2 = str(tmp_path / f"m2_{name}.{ext}")
        bst2.save_model(p2)
        with open(path, "rb") as a, open(p2, "rb") as b:
            assert a.read() == b.read()

    def test_ubj_is_binary_not_json(self, tmp_path):
        bst, _ = _train()
        path = str(tmp_path / "m.ubj")
        bst.save_model(path)
        raw = open(path, "rb").read()
        assert raw[:1] == b"{"
        # UBJSON object: first key is length-prefixed with an int marker
        assert raw[1:2] in b"iUIlL"
        with pytest.raises(json.JSONDecodeError):
            json.loads(raw.decode("latin-1"))
        # and the xgb spelling matches byte-for-byte
        p2 = str(tmp_path / "m.xgb")
        bst.save_model(p2)
        assert open(p2, "rb").read() == raw

    def test_ubj_json_same_semantics(self, tmp_path):
        bst, X = _train()
        jraw = bst.save_raw("json")
        uraw = bst.save_raw("ubj")
        assert jraw[:1] == b"{" and uraw[1:2] in b"iUIlL"
        b_j = Booster().load_model(jraw)
        b_u = Booster().load_model(uraw)
        np.testing.assert_array_equal(b_j.predict(X), b_u.predict(X))

    def test_save_raw_rejects_deprecated(self):
        bst, _ = _train(rounds=1)
        with pytest.raises(ValueError):
            bst.save_raw("deprecated")

    def test_load_model_sniffs_bytes(self):
        bst, X = _train()
        for raw in (bst.save_raw("json"), bst.save_raw("ubj")):
            b2 = Booster().load_model(raw)
            np.testing.assert_array_equal(bst.predict(X), b2.predict(X))


class TestGoldenModel:
    """A vendored stock-XGBoost-schema model: predictions computed by
    hand from the file, and load->save->load->save byte stability."""

    def _hand_margin(self, x):
        # tree 0: f0 < 1.0 (default left) -> -0.4 / 0.6
        # tree 1: f1 < 0.0 (default right) -> 0.25 / -0.15
        t0 = -0.4 if (np.isnan(x[0]) or x[0] < 1.0) else 0.6
        if np.isnan(x[1]):
            t1 = -0.15
        else:
            t1 = 0.25 if x[1] < 0.0 else -0.15
        return t0 + t1  # base_score 0.5 -> margin 0

    def test_golden_predictions(self):
        bst = Booster().load_model(GOLDEN)
        X = np.array(
            [[0.0, -1.0], [2.0, 1.0], [np.nan, np.nan], [1.0, 0.0]],
            dtype=np.float32,
        )
        want = np.array([self._hand_margin(r) for r in X], np.float32)
        got = bst.predict(X, output_margin=True)
        np.testing.assert_allclose(got, want, rtol=1e-6)
        p = bst.predict(X)
        np.testing.assert_allclose(
            p, 1.0 / (1.0 + np.exp(-want)), rtol=1e-6
        )

    def test_golden_fields_survive_roundtrip(self, tmp_path):
        bst = Booster().load_model(GOLDEN)
        out = str(tmp_path / "g.json")
        bst.save_model(out)
        golden = json.load(open(GOLDEN))
        ours = json.load(open(out))
        gl, ol = golden["learner"], ours["learner"]
        assert ol["objective"]["name"] == gl["objective"]["name"]
        glp, olp = gl["learner_model_param"], ol["learner_model_param"]
        assert float(olp["base_score"]) == float(glp["base_score"])
        assert olp["num_feature"] == glp["num_feature"]
        assert olp["num_class"] == glp["num_class"]
        gm = gl["gradient_booster"]["model"]
        om = ol["gradient_booster"]["model"]
        assert om["tree_info"] == gm["tree_info"]
        assert om["iteration_indptr"] == gm["iteration_indptr"]
        assert (om["gbtree_model_param"]["num_trees"]
                == gm["gbtree_model_param"]["num_trees"])
        for gt, ot in zip(gm["trees"], om["trees"]):
            for key in ("left_children", "right_children", "parents",
                        "split_indices", "default_left", "split_type"):
                assert ot[key] == gt[key], key
            for key in ("split_conditions", "base_weights",
                        "loss_changes", "sum_hessian"):
                np.testing.assert_allclose(ot[key], gt[key], rtol=1e-6,
                                           err_msg=key)
            assert (ot["tree_param"]["num_nodes"]
                    == gt["tree_param"]["num_nodes"])

    @pytest.mark.parametrize("ext", ["json", "ubj"])
    def test_golden_generation_stability(self, tmp_path, ext):
        bst = Booster().load_model(GOLDEN)
        a = str(tmp_path / f"a.{ext}")
        b = str(tmp_path / f"b.{ext}")
        bst.save_model(a)
        Booster().load_model(a).save_model(b)
        assert open(a, "rb").read() == open(b, "rb").read()
