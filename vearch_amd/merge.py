"""Cross-partition top-k merge — the router's semantics
(reference: internal/client/client.go:1497 mergeSortedArrays /
:1558 AddMergeSort): each partition returns its top-k sorted by score;
the router k-way merges into the global top-k. Deterministic ties by
(score, partition-global id).

Vectorized via the same u64 (dist-key | id) packing the GPU selector
uses: one np.sort per batch (this runs once per query batch on rank 0
of the 8-GPU bench, so it must not gate the step)."""
import numpy as np


def _f32_key(d):
    """monotone u32 image of fp32 (sign-flip trick, select.hpp)."""
    u = d.astype(np.float32).view(np.uint32)
    neg = (u & 0x80000000) != 0
    return np.where(neg, ~u, u | np.uint32(0x80000000)).astype(np.uint64)


def merge_topk(dists_list, ids_list, k, descending=False):
    """dists_list/ids_list: per-partition (nq, k_i) arrays with docids
    already globalized (-1 = empty slot, ids < 2^32). Returns (nq, k)
    merged arrays (-1 padded)."""
    dists = np.ascontiguousarray(
        np.concatenate(dists_list, axis=1), dtype=np.float32)
    ids = np.concatenate(ids_list, axis=1).astype(np.int64)
    dk = _f32_key(dists)
    if descending:
        dk = np.uint64(0xFFFFFFFF) - dk  # invert: ascending == best-first
    key = (dk << np.uint64(32)) | ids.astype(np.uint64)
    key[ids < 0] = np.uint64(0xFFFFFFFFFFFFFFFF)  # empties sort last
    key = np.sort(key, axis=1)[:, :k]
    empty = key == np.uint64(0xFFFFFFFFFFFFFFFF)
    out_i = np.where(empty, -1, (key & np.uint64(0xFFFFFFFF)).astype(np.int64))
    dk = (key >> np.uint64(32)).astype(np.uint32)
    if descending:
        dk = np.uint32(0xFFFFFFFF) - dk
    neg = (dk & 0x80000000) == 0
    u = np.where(neg, ~dk, dk & np.uint32(0x7FFFFFFF)).astype(np.uint32)
    out_d = np.where(empty, np.float32(-1.0), u.view(np.float32))
    if out_d.shape[1] < k:  # fewer candidates than k: pad
        pad = k - out_d.shape[1]
        out_d = np.pad(out_d, ((0, 0), (0, pad)), constant_values=-1.0)
        out_i = np.pad(out_i, ((0, 0), (0, pad)), constant_values=-1)
    return np.ascontiguousarray(out_d), np.ascontiguousarray(out_i)


def pack_keys_signed(dists, ids, descending=False):
    """(nq,k) -> int64 keys whose SIGNED order == (dist, id) order
    (top bit biased), for torch/all_gather transport. -1 ids map to
    int64 max."""
    dk = _f32_key(dists)
    if descending:
        dk = np.uint64(0xFFFFFFFF) - dk
    skey = ((dk.astype(np.int64) - (1 << 31)) << 32) |         (ids & np.int64(0xFFFFFFFF))
    skey[ids < 0] = np.iinfo(np.int64).max
    return skey


def unpack_keys_signed(skey, descending=False):
    empty = skey == np.iinfo(np.int64).max
    ids = np.where(empty, -1, skey & np.int64(0xFFFFFFFF))
    dk = ((skey >> 32) + (1 << 31)).astype(np.uint32)
    if descending:
        dk = np.uint32(0xFFFFFFFF) - dk
    neg = (dk & 0x80000000) == 0
    u = np.where(neg, ~dk, dk & np.uint32(0x7FFFFFFF)).astype(np.uint32)
    dists = np.where(empty, np.float32(-1.0), u.view(np.float32))
    return dists, ids


def pack_keys_signed_torch(dists_t, ids_t, world=1, rank=0,
                           descending=False):
    """torch version of pack_keys_signed with round-robin id
    globalization fused (runs on GPU in the multi-rank bench so packing
    never gates the step). dists_t float32, ids_t int64 tensors."""
    import torch
    u = dists_t.view(torch.int32).to(torch.int64) & 0xFFFFFFFF
    neg = (u >> 31) != 0
    dk = torch.where(neg, (~u) & 0xFFFFFFFF, u | 0x80000000)
    if descending:
        dk = 0xFFFFFFFF - dk
    gids = ids_t * world + rank
    skey = ((dk - (1 << 31)) << 32) | (gids & 0xFFFFFFFF)
    return torch.where(ids_t < 0,
                       torch.tensor(torch.iinfo(torch.int64).max,
                                    dtype=torch.int64,
                                    device=dists_t.device), skey)
