"""Round trip: tools/convert_dataset.from_arrays writes the reference's
four on-disk formats (generate_nts_dataset.py:162-226) and this repo's
loaders (graph.load_gemini_edges, data.read_*) read them back exactly.
The DGL/OGB adapters are import-gated (neither library exists in this
image); the array path is the tested product."""
import sys

import numpy as np
import pytest

from neutronstarlite_amd import data as D
from neutronstarlite_amd import graph as G

sys.path.insert(0, "tools")
from convert_dataset import add_self_loops, from_arrays  # noqa: E402


def test_round_trip(tmp_path):
    v, f, e = 37, 6, 150
    rng = np.random.default_rng(5)
    edges = np.stack([rng.integers(0, v, e), rng.integers(0, v, e)],
                     axis=1).astype(np.uint32)
    feats = rng.normal(size=(v, f)).astype(np.float32)
    labels = rng.integers(0, 4, v)
    train = np.zeros(v, bool); train[:10] = True
    val = np.zeros(v, bool); val[10:20] = True
    test = np.zeros(v, bool); test[20:25] = True
    prefix = str(tmp_path / "toy")
    from_arrays(prefix, edges, feats, labels, train, val, test,
                self_loop=True)

    back = G.load_gemini_edges(prefix + ".edge")
    expect = add_self_loops(edges, v)
    assert np.array_equal(back, expect)
    # self-loop convention: no duplicate self loops, one per vertex
    loops = back[back[:, 0] == back[:, 1]]
    assert np.array_equal(np.sort(loops[:, 0]), np.arange(v))

    x = D.read_feature_table(prefix + ".feat", 0, v, f)
    assert np.allclose(x, feats, atol=5e-5)  # %.4f quantization
    y = D.read_label_table(prefix + ".label", 0, v)
    assert np.array_equal(y, labels)
    m = D.read_mask(prefix + ".mask", 0, v)
    # data.read_mask encoding: check the train/val/test split survives
    assert (m[:10] == m[0]).all() and (m[10:20] == m[10]).all()
    assert len({int(m[0]), int(m[10]), int(m[20]), int(m[30])}) >= 3


def test_adapters_gated():
    from convert_dataset import from_dgl, from_ogb
    with pytest.raises(SystemExit):
        from_dgl("cora", ".")
    with pytest.raises(SystemExit):
        from_ogb("ogbn-arxiv", ".")
