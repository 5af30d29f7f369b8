"""LibSVM text format reader (reference behavior: dmlc-core text
parsers used by DMatrix::Load; text input is deprecated upstream but
still accepted — data.cc:930).  Supports 'label idx:val ...' lines and
optional 'qid:' tokens."""
from __future__ import annotations

from typing import Optional

import numpy as np


def load_svmlight(path: str, n_features: Optional[int] = None,
                  zero_based: bool = True):
    """Returns (X dense float32 [n, f], y float32 [n], qid or None).

    Missing entries are 0 (libsvm sparse semantics)."""
    labels = []
    qids = []
    rows = []
    max_idx = -1
    with open(path) as fh:
        for line in fh:
            line = line.strip()
            if not line or line.startswith("#"):
                continue
            parts = line.split()
            labels.append(float(parts[0]))
            feats = []
            for tok in parts[1:]:
                if tok.startswith("qid:"):
                    qids.append(int(tok[4:]))
                    continue
                idx, val = tok.split(":")
                i = int(idx)
                feats.append((i, float(val)))
                max_idx = max(max_idx, i)
            rows.append(feats)
    if n_features is None:
        n_features = max_idx + 1
    X = np.zeros((len(rows), n_features), dtype=np.float32)
    for r, feats in enumerate(rows):
        for i, v in feats:
            if i < n_features:
                X[r, i] = v
    y = np.asarray(labels, dtype=np.float32)
    q = np.asarray(qids, dtype=np.int64) if qids else None
    return X, y, q
