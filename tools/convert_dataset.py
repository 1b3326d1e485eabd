#!/usr/bin/env python3
"""Dataset converter to the NTS on-disk formats.

Produces exactly the four files the reference's exporter writes
(/root/reference/data/generate_nts_dataset.py:162-226, formats also
described in data/reddit/note_for_input.txt) and that this repo's loaders
read (neutronstarlite_amd/data.py, graph.load_gemini_edges):

  <prefix>.edge   Gemini binary: consecutive (src, dst) u32 pairs, 8 B/edge,
                  native byte order
  <prefix>.feat   text, "idx f0 f1 ... fD-1" per vertex, %.4f
  <prefix>.label  text, "idx label" per vertex
  <prefix>.mask   text, "idx train|val|test|unknown" per vertex

Two entry points:
  - from_arrays(...): pure-numpy, runs anywhere (unit-tested in
    tests/test_convert_dataset.py)
  - from_dgl(name)/from_ogb(name): thin adapters that import dgl/ogb
    LAZILY; this build image has neither and no network, so they raise a
    clear error here — the array path is the tested product, the adapters
    are the glue a user runs where DGL exists (mirroring the reference's
    extract_dataset, generate_nts_dataset.py:21-143).

Self-loop handling matches the reference (--self-loop default True:
remove existing self loops, then add one per vertex).
"""
import argparse
import os
import sys

import numpy as np


def add_self_loops(edges: np.ndarray, v: int) -> np.ndarray:
    """remove_self_loop + add_self_loop (generate_nts_dataset.py:41-47)."""
    edges = edges[edges[:, 0] != edges[:, 1]]
    loops = np.stack([np.arange(v, dtype=edges.dtype)] * 2, axis=1)
    return np.concatenate([edges, loops], axis=0)


def from_arrays(out_prefix: str, edges: np.ndarray, features: np.ndarray,
                labels: np.ndarray, train_mask: np.ndarray,
                val_mask: np.ndarray, test_mask: np.ndarray,
                self_loop: bool = True) -> None:
    """Write the four NTS files from plain arrays.

    edges: (E,2) int (src,dst); features: (V,F) float; labels: (V,) int;
    masks: (V,) bool each (train wins over val over test, reference
    precedence, generate_nts_dataset.py:180-189)."""
    v = features.shape[0]
    assert labels.shape[0] == v and train_mask.shape[0] == v
    edges = np.asarray(edges, dtype=np.uint32)
    if self_loop:
        edges = add_self_loops(edges, v)
    # .edge — Gemini binary u32 pairs (edge2bin, :194-199)
    edges.astype(np.uint32).tofile(out_prefix + ".edge")
    # .feat — "idx f0 ... fD-1", %.4f (write_to_file(index=True), :209-224)
    with open(out_prefix + ".feat", "w") as f:
        for i in range(v):
            f.write(str(i) + " " +
                    " ".join("%.4f" % x for x in features[i]) + "\n")
    # .label — "idx label"
    with open(out_prefix + ".label", "w") as f:
        for i in range(v):
            f.write(f"{i} {int(labels[i])}\n")
    # .mask — "idx train|val|test|unknown" (write_to_mask, :200-207)
    with open(out_prefix + ".mask", "w") as f:
        for i in range(v):
            if train_mask[i]:
                kind = "train"
            elif val_mask[i]:
                kind = "val"
            elif test_mask[i]:
                kind = "test"
            else:
                kind = "unknown"
            f.write(f"{i} {kind}\n")


def from_dgl(name: str, out_dir: str = ".", self_loop: bool = True) -> None:
    """cora/citeseer/pubmed/reddit via DGL (needs dgl installed + network)."""
    try:
        import dgl  # noqa: F401
        from dgl.data import load_data
    except ImportError as exc:  # pragma: no cover - no dgl in this image
        raise SystemExit(
            "dgl is not installed in this environment; run this converter "
            "where DGL and the dataset downloads are available") from exc
    ns = argparse.Namespace(dataset=name)
    data = load_data(ns)
    g = data[0]
    if self_loop:
        g = dgl.add_self_loop(dgl.remove_self_loop(g))
    src, dst = (t.numpy() for t in g.edges())
    from_arrays(os.path.join(out_dir, name),
                np.stack([src, dst], axis=1),
                g.ndata["feat"].numpy(), g.ndata["label"].numpy(),
                g.ndata["train_mask"].numpy(), g.ndata["val_mask"].numpy(),
                g.ndata["test_mask"].numpy(), self_loop=False)


def from_ogb(name: str, out_dir: str = ".", self_loop: bool = True) -> None:
    """ogbn-* via OGB (needs ogb installed + network)."""
    try:
        from ogb.nodeproppred import DglNodePropPredDataset
    except ImportError as exc:  # pragma: no cover - no ogb in this image
        raise SystemExit(
            "ogb is not installed in this environment; run this converter "
            "where OGB and the dataset downloads are available") from exc
    data = DglNodePropPredDataset(name=name)
    g, labels = data[0]
    split = data.get_idx_split()
    v = g.number_of_nodes()
    masks = {}
    for k in ("train", "valid", "test"):
        m = np.zeros(v, dtype=bool)
        m[split[k].numpy()] = True
        masks[k] = m
    src, dst = (t.numpy() for t in g.edges())
    from_arrays(os.path.join(out_dir, name.replace("-", "_")),
                np.stack([src, dst], axis=1), g.ndata["feat"].numpy(),
                labels.numpy().reshape(-1), masks["train"], masks["valid"],
                masks["test"], self_loop=self_loop)


if __name__ == "__main__":
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--dataset", default="cora")
    ap.add_argument("--out-dir", default=".")
    ap.add_argument("--no-self-loop", action="store_true")
    args = ap.parse_args()
    if args.dataset.startswith("ogbn-"):
        from_ogb(args.dataset, args.out_dir, not args.no_self_loop)
    else:
        from_dgl(args.dataset, args.out_dir, not args.no_self_loop)
    print("wrote", args.dataset, "to", args.out_dir, file=sys.stderr)
