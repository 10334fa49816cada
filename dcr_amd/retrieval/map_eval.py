"""Oxford/Paris-style retrieval mAP (capability parity with
/root/reference/utils_ret.py:300-417 compute_map — kept because the
reference library exposes it, even though the DCR pipelines don't call
it in their main paths)."""
from __future__ import annotations

from typing import Dict, Sequence

import numpy as np


def compute_ap(ranks: np.ndarray, nres: int) -> float:
    """Average precision for one query given 0-indexed positions of
    positive images in the ranked list."""
    if nres == 0:
        return 0.0
    ap = 0.0
    recall_step = 1.0 / nres
    for j, rank in enumerate(ranks):
        precision_0 = j / rank if rank > 0 else 1.0
        precision_1 = (j + 1) / (rank + 1)
        ap += (precision_0 + precision_1) / 2.0 * recall_step
    return ap


def compute_map(ranks: np.ndarray, gnd: Sequence[Dict], kappas: Sequence[int] = ()):
    """ranks: [n_db, n_query] ranked db indices per query column;
    gnd[i]: {'ok': positive db ids, 'junk': ignored db ids}.
    Returns (mAP, per-query aps, mpr@k, per-query pr@k)."""
    n_q = ranks.shape[1]
    aps = np.zeros(n_q)
    prs = np.zeros((n_q, len(kappas)))
    n_empty = 0
    for i in range(n_q):
        ok = np.asarray(gnd[i].get("ok", []))
        junk = np.asarray(gnd[i].get("junk", []))
        if ok.size == 0:
            aps[i] = float("nan")
            prs[i, :] = float("nan")
            n_empty += 1
            continue
        col = ranks[:, i]
        pos = np.in1d(col, ok).nonzero()[0]
        jk = np.in1d(col, junk).nonzero()[0]
        if jk.size:
            # shift positive ranks down by the junk entries above them
            shift = np.searchsorted(jk, pos)
            pos = pos - shift
        aps[i] = compute_ap(pos, ok.size)
        for k, kappa in enumerate(kappas):
            prs[i, k] = (pos < kappa).sum() / min(kappa, ok.size)
    valid = max(n_q - n_empty, 1)
    map_ = float(np.nansum(aps) / valid)
    mpr = np.nansum(prs, axis=0) / valid if len(kappas) else np.array([])
    return map_, aps, mpr, prs
