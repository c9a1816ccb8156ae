"""Cross-partition top-k merge — the router's semantics
(reference: internal/client/client.go:1497 mergeSortedArrays /
:1558 AddMergeSort): each partition returns its top-k sorted by score;
the router k-way merges into the global top-k. Deterministic ties by
(score, partition-global id)."""
import numpy as np


def merge_topk(dists_list, ids_list, k, descending=False):
    """dists_list/ids_list: per-partition (nq, k_i) arrays with docids
    already globalized. Returns (nq, k) merged arrays (-1 padded)."""
    dists = np.concatenate(dists_list, axis=1)
    ids = np.concatenate(ids_list, axis=1)
    nq = dists.shape[0]
    out_d = np.full((nq, k), -1.0, dtype=np.float32)
    out_i = np.full((nq, k), -1, dtype=np.int64)
    for i in range(nq):
        valid = ids[i] >= 0
        dv, iv = dists[i][valid], ids[i][valid]
        key = -dv if descending else dv
        order = np.lexsort((iv, key))[:k]
        out_d[i, :len(order)] = dv[order]
        out_i[i, :len(order)] = iv[order]
    return out_d, out_i
