"""On-disk dataset ingestion (pre-downloaded real datasets).

The reference loads Reddit/Yelp through dgl.data and ogbn-products/
papers100M through ogb (reference: helper/utils.py:21-70) — both
download over the network. This environment has no egress, so this
loader consumes data ALREADY ON DISK in a documented layout and applies
the same post-processing the reference does:

* self-loops removed and re-added (helper/utils.py:67-69);
* Yelp: labels cast to float32 and a StandardScaler fit on the TRAIN
  rows' features is applied to ALL features (helper/utils.py:53-57);
* n_feat/n_class inferred from the arrays; multilabel iff the label
  array is 2-D (helper/utils.py:62-65).

Layout: one file `<data_path>/<dataset>.npz` with keys
  src, dst            int32/int64 [E]      directed edges (src -> dst), OR
  indptr, indices     CSR keyed by dst (row = destination)
  feat                float32 [N, F]
  label               int [N] (single-label) or {0,1} [N, C] (multilabel)
  train_mask, val_mask, test_mask   bool [N]

Activated by `--dataset <name> --data-path <dir>` whenever that file
exists; otherwise the synthetic generator of the named shape is used
(graph/synthetic.py).
"""
from __future__ import annotations

import os

import numpy as np

from .csr import CSR, Graph, add_self_loops


def disk_dataset_file(name: str, data_path: str | None) -> str | None:
    if not data_path:
        return None
    f = os.path.join(data_path, f"{name}.npz")
    return f if os.path.exists(f) else None


def load_disk_data(name: str, data_path: str) -> Graph:
    f = disk_dataset_file(name, data_path)
    if f is None:
        raise FileNotFoundError(
            f"no on-disk dataset {name!r} under {data_path!r} "
            f"(expected {name}.npz — see graph/ingest.py for the layout)")
    z = np.load(f)
    feat = np.ascontiguousarray(z["feat"], dtype=np.float32)
    n_nodes = feat.shape[0]
    label = z["label"]

    if "indptr" in z:
        csr = CSR(z["indptr"], z["indices"], n_nodes)
        src, dst = csr.to_edges()
    else:
        src, dst = np.asarray(z["src"]), np.asarray(z["dst"])
    # reference post-processing: drop + re-add self-loops (utils.py:67-69)
    src, dst = add_self_loops(src.astype(np.int64), dst.astype(np.int64),
                              n_nodes)
    adj_in = CSR.from_edges(src, dst, n_nodes, n_nodes, sort_cols=True)

    train_mask = np.asarray(z["train_mask"]).astype(bool)
    val_mask = np.asarray(z["val_mask"]).astype(bool)
    test_mask = np.asarray(z["test_mask"]).astype(bool)

    multilabel = label.ndim > 1                  # utils.py:62-65
    if multilabel:
        label = label.astype(np.float32)
        n_class = label.shape[1]
    else:
        label = label.astype(np.int64)
        n_class = int(label.max()) + 1

    if name == "yelp":
        # float labels + train-fit StandardScaler on features
        # (reference utils.py:53-57)
        from sklearn.preprocessing import StandardScaler
        label = label.astype(np.float32)
        scaler = StandardScaler()
        scaler.fit(feat[train_mask])
        feat = scaler.transform(feat).astype(np.float32)

    return Graph(adj_in, feat, label, train_mask, val_mask, test_mask,
                 n_class, multilabel, name=name)
