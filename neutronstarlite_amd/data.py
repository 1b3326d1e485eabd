"""GNNDatum-equivalent data loading (SURVEY §2 "KEEP minimal"): the
reference's text formats for features, labels and masks, plus its synthetic
convention (all citations into /root/reference/core/ntsDataloador.hpp):

  - feature table: lines "id f0 f1 ... f_{F-1}" (readFeature_Label_Mask,
    :156-221; whitespace-separated, id first);
  - label table:  lines "id label" (:190-191);
  - mask file:    lines "id {train|eval|val|test}" -> 0/1/2, anything else 3
    (:195-205);
  - random_generate (:63-71): every feature 1.0, label = rand() % classes,
    mask = i % 3 — the deterministic all-ones convention our closed-form
    parity checks reuse.

Vectorized numpy; rows outside [v_start, v_end) are skipped like the
reference's partition filter (:145-152).
"""
import numpy as np


def read_feature_table(path: str, v_start: int, v_end: int,
                       f: int) -> np.ndarray:
    """Text features -> fp32 [v_end-v_start, f]."""
    raw = np.loadtxt(path, dtype=np.float64, ndmin=2)
    assert raw.shape[1] == f + 1, (
        f"feature table width {raw.shape[1]} != id + {f}")
    ids = raw[:, 0].astype(np.int64)
    out = np.zeros((v_end - v_start, f), dtype=np.float32)
    sel = (ids >= v_start) & (ids < v_end)
    out[ids[sel] - v_start] = raw[sel, 1:].astype(np.float32)
    return out


def read_label_table(path: str, v_start: int, v_end: int) -> np.ndarray:
    """Text labels -> int64 [v_end-v_start]."""
    raw = np.loadtxt(path, dtype=np.int64, ndmin=2)
    ids = raw[:, 0]
    out = np.zeros(v_end - v_start, dtype=np.int64)
    sel = (ids >= v_start) & (ids < v_end)
    out[ids[sel] - v_start] = raw[sel, 1]
    return out


_MASK_CODE = {"train": 0, "eval": 1, "val": 1, "test": 2}


def read_mask(path: str, v_start: int, v_end: int) -> np.ndarray:
    """Text masks -> int32 [v_end-v_start]: train=0 eval/val=1 test=2 else 3."""
    out = np.full(v_end - v_start, 3, dtype=np.int32)
    with open(path) as fh:
        for line in fh:
            parts = line.split()
            if len(parts) < 2:
                continue
            vid = int(parts[0])
            if v_start <= vid < v_end:
                out[vid - v_start] = _MASK_CODE.get(parts[1], 3)
    return out


def random_generate(n: int, f: int, classes: int, seed: int = 0):
    """The reference's deterministic synthetic convention
    (ntsDataloador.hpp:63-71): features all ones, labels uniform in
    [0, classes), mask = i % 3."""
    rng = np.random.default_rng(seed)
    features = np.ones((n, f), dtype=np.float32)
    labels = rng.integers(0, classes, size=n).astype(np.int64)
    mask = (np.arange(n) % 3).astype(np.int32)
    return features, labels, mask
