"""Cross-partition top-k merge — the router's semantics
(reference: internal/client/client.go:1497 mergeSortedArrays /
:1558 AddMergeSort): each partition returns its top-k sorted by score;
the router k-way merges into the global top-k. Deterministic ties by
(score, partition-global id). Fully vectorized (this runs once per
query batch on rank 0 of the 8-GPU bench)."""
import numpy as np


def merge_topk(dists_list, ids_list, k, descending=False):
    """dists_list/ids_list: per-partition (nq, k_i) arrays with docids
    already globalized (-1 = empty slot). Returns (nq, k) merged arrays
    (-1 padded)."""
    dists = np.concatenate(dists_list, axis=1).astype(np.float32)
    ids = np.concatenate(ids_list, axis=1)
    invalid = ids < 0
    key = -dists if descending else dists.copy()
    key[invalid] = np.inf
    # lexsort over the last axis: primary key dist, secondary id
    order = np.lexsort((np.where(invalid, np.iinfo(np.int64).max, ids),
                        key), axis=1)[:, :k]
    out_d = np.take_along_axis(dists, order, 1)
    out_i = np.take_along_axis(ids, order, 1)
    inv = np.take_along_axis(invalid, order, 1)
    out_d = np.where(inv, np.float32(-1.0), out_d)
    out_i = np.where(inv, np.int64(-1), out_i)
    if out_d.shape[1] < k:  # fewer candidates than k: pad
        pad = k - out_d.shape[1]
        out_d = np.pad(out_d, ((0, 0), (0, pad)), constant_values=-1.0)
        out_i = np.pad(out_i, ((0, 0), (0, pad)), constant_values=-1)
    return np.ascontiguousarray(out_d), np.ascontiguousarray(out_i)
