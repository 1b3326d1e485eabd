"""GNNDatum-equivalent text loaders (feature/label/mask formats of
core/ntsDataloador.hpp) — round-trip and partition-filter behavior, plus a
check against the vendored cora.labeltable/mask when the reference is
mounted (build container only)."""
import os

import numpy as np
import pytest

from neutronstarlite_amd import data as D
from tests.conftest import REFERENCE


def test_text_roundtrip_and_partition_filter(tmp_path):
    v, f = 10, 3
    rng = np.random.default_rng(0)
    feats = rng.normal(size=(v, f)).astype(np.float32)
    labels = rng.integers(0, 4, size=v)
    ftr = tmp_path / "x.featuretable"
    lbl = tmp_path / "x.labeltable"
    msk = tmp_path / "x.mask"
    names = ["train", "eval", "test", "val", "other"]
    with open(ftr, "w") as a, open(lbl, "w") as b, open(msk, "w") as c:
        for i in range(v):
            a.write(f"{i} " + " ".join(f"{x:.6f}" for x in feats[i]) + "\n")
            b.write(f"{i} {labels[i]}\n")
            c.write(f"{i} {names[i % 5]}\n")
    got = D.read_feature_table(str(ftr), 0, v, f)
    assert np.allclose(got, feats, atol=1e-6)
    assert np.array_equal(D.read_label_table(str(lbl), 0, v), labels)
    m = D.read_mask(str(msk), 0, v)
    assert m.tolist() == [0, 1, 2, 1, 3, 0, 1, 2, 1, 3]
    # partition filter: only [4, 8) kept, local ids
    part = D.read_feature_table(str(ftr), 4, 8, f)
    assert np.allclose(part, feats[4:8], atol=1e-6)


def test_random_generate_convention():
    x, y, m = D.random_generate(9, 4, classes=7, seed=1)
    assert np.all(x == 1.0)           # ntsDataloador.hpp:66 all-ones
    assert y.min() >= 0 and y.max() < 7
    assert m.tolist() == [0, 1, 2, 0, 1, 2, 0, 1, 2]


@pytest.mark.skipif(not os.path.exists(REFERENCE),
                    reason="reference not mounted (run-time box)")
def test_vendored_cora_label_and_mask_parse():
    labels = D.read_label_table(
        os.path.join(REFERENCE, "data", "cora.labeltable"), 0, 2708)
    assert labels.shape == (2708,)
    assert labels.min() >= 0 and labels.max() <= 6  # 7 Cora classes
    mask = D.read_mask(os.path.join(REFERENCE, "data", "cora.mask"), 0, 2708)
    assert set(np.unique(mask)).issubset({0, 1, 2, 3})
    assert (mask == 0).sum() > 0 and (mask == 2).sum() > 0
