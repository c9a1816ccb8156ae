"""Multi-partition path on CPU: world_size=2 over gloo exercises the same
all-gather + router-merge (client.go:1497/1558 semantics) that bench.py
runs over RCCL on the 8-GPU node. Partition search is played by the CPU
oracle; the collective and merge code are the real ones."""
import os

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

import oracle as orc
from vearch_amd.merge import merge_topk

K = 10


def _worker(rank, world, port, out_q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    base = orc.gen_clustered(4000, 32, seed=42, ncl=50)
    q = orc.gen_queries(base, 16, seed=3)
    try:
        _worker_body(rank, world, base, q, out_q)
    finally:
        dist.destroy_process_group()


def _worker_body(rank, world, base, q, out_q):
    # round-robin sharding: rank owns rows where i % world == rank
    my_rows = np.arange(rank, 4000, world)
    my_base = base[my_rows]
    d_loc, i_loc = orc.flat_search(my_base, q, K, "L2")
    # globalize local ids
    gi = np.where(i_loc >= 0, my_rows[np.clip(i_loc, 0, None)], -1)
    payload = torch.from_numpy(
        np.concatenate([d_loc.astype(np.float32).reshape(16, K, 1),
                        gi.astype(np.float32).reshape(16, K, 1)], axis=2))
    gathered = [torch.empty_like(payload) for _ in range(world)]
    dist.all_gather(gathered, payload)
    if rank == 0:
        dl = [g[:, :, 0].numpy() for g in gathered]
        il = [g[:, :, 1].numpy().astype(np.int64) for g in gathered]
        md, mi = merge_topk(dl, il, K)
        od, oi = orc.flat_search(base, q, K, "L2")
        out_q.put((np.array_equal(mi, oi), np.allclose(md, od)))


@pytest.mark.parametrize("world,port", [(2, 29611), (4, 29617)])
def test_partition_merge_equals_global(world, port):
    """world partitions (2 and 4 — the 8-GPU merge is the same code
    with a bigger gather) must merge to the global FLAT result."""
    ctx = mp.get_context("spawn")
    out_q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, world, port, out_q))
             for r in range(world)]
    for p in procs:
        p.start()
    ok_ids, ok_dists = out_q.get(timeout=300)
    for p in procs:
        p.join(timeout=120)
    assert ok_ids, "merged ids != global FLAT ids"
    assert ok_dists, "merged dists != global FLAT dists"


def test_merge_topk_ties_and_padding():
    d1 = np.array([[0.5, 1.0, -1.0]], dtype=np.float32)
    i1 = np.array([[7, 3, -1]], dtype=np.int64)
    d2 = np.array([[0.5, 2.0, -1.0]], dtype=np.float32)
    i2 = np.array([[4, 9, -1]], dtype=np.int64)
    md, mi = merge_topk([d1, d2], [i1, i2], 4)
    assert mi[0].tolist() == [4, 7, 3, 9]  # tie 0.5 -> lower id first
    md, mi = merge_topk([d1, d2], [i1, i2], 6)
    assert mi[0].tolist() == [4, 7, 3, 9, -1, -1]


def test_packed_key_merge_matches_reference_merge():
    """The transport form bench.py uses (signed-int64 packed keys sorted
    by torch) must merge identically to merge_topk."""
    import numpy as np
    from vearch_amd.merge import (merge_topk, pack_keys_signed,
                                  unpack_keys_signed)
    rng = np.random.default_rng(3)
    for descending in (False, True):
        dl = [rng.standard_normal((50, 7)).astype(np.float32)
              for _ in range(3)]
        il = [rng.integers(0, 10**6, (50, 7)).astype(np.int64)
              for _ in range(3)]
        il[1][:, 5:] = -1  # padding slots
        want_d, want_i = merge_topk(dl, il, 9, descending=descending)
        keys = np.concatenate(
            [pack_keys_signed(d, i, descending) for d, i in zip(dl, il)],
            axis=1)
        t = torch.from_numpy(keys)
        merged = torch.sort(t, dim=1).values[:, :9].numpy()
        got_d, got_i = unpack_keys_signed(merged, descending)
        assert np.array_equal(got_i, want_i)
        assert np.array_equal(got_d, want_d)


def test_torch_pack_matches_numpy_pack():
    import numpy as np
    from vearch_amd.merge import pack_keys_signed, pack_keys_signed_torch
    rng = np.random.default_rng(7)
    d = rng.standard_normal((20, 5)).astype(np.float32)
    i = rng.integers(0, 10**6, (20, 5)).astype(np.int64)
    i[3, 2:] = -1
    for world, rank in ((1, 0), (4, 3)):
        gids = np.where(i >= 0, i * world + rank, -1)
        want = pack_keys_signed(d, gids)
        got = pack_keys_signed_torch(torch.from_numpy(d),
                                     torch.from_numpy(i), world,
                                     rank).numpy()
        assert np.array_equal(want, got)


def test_merge_topk_property_random():
    """Property check over random shapes: merge_topk == brute-force
    concatenate+stable-sort by (dist, id), empties last, -1 padding;
    both metric orders."""
    rng = np.random.default_rng(42)
    for trial in range(200):
        nq = int(rng.integers(1, 8))
        nparts = int(rng.integers(1, 5))
        k = int(rng.integers(1, 12))
        descending = bool(rng.integers(0, 2))
        dl, il = [], []
        for _ in range(nparts):
            kp = int(rng.integers(0, 9))
            d = rng.standard_normal((nq, kp)).astype(np.float32)
            # inject exact ties across partitions
            d[rng.random((nq, kp)) < 0.3] = np.float32(0.25)
            i = rng.integers(0, 5000, (nq, kp)).astype(np.int64)
            i[rng.random((nq, kp)) < 0.2] = -1  # empty slots
            # partition results arrive sorted best-first like the PS
            key = np.where(i < 0, np.inf,
                           -d if descending else d)
            order = np.argsort(key, axis=1, kind="stable")
            dl.append(np.take_along_axis(d, order, 1))
            il.append(np.take_along_axis(i, order, 1))
        md, mi = merge_topk(dl, il, k, descending=descending)
        assert md.shape == (nq, k) and mi.shape == (nq, k)
        for t in range(nq):
            cand = []
            for d, i in zip(dl, il):
                for j in range(d.shape[1]):
                    if i[t, j] >= 0:
                        cand.append((float(d[t, j]), int(i[t, j])))
            cand.sort(key=lambda p: (-p[0] if descending else p[0], p[1]))
            want = cand[:k]
            got = [(float(md[t, j]), int(mi[t, j]))
                   for j in range(k) if mi[t, j] >= 0]
            assert got == want, (trial, t, got, want)
            assert all(mi[t, j] == -1 for j in range(len(want), k))


def test_f32_key_total_order_extremes():
    """The u64 (dist-key | id) packing (select.hpp sign-flip trick,
    mirrored in merge._f32_key) must be a strict monotone image of fp32
    over the full finite+inf range, including denormals and max-mag
    values. (+-0.0 note: the key image distinguishes -0.0 < +0.0;
    computed L2^2/IP distances are never -0.0 — sums of products
    rounding to zero give +0.0 under IEEE round-to-nearest — so this
    never diverges from the oracle's float compare.)"""
    from vearch_amd.merge import _f32_key
    vals = np.array([-np.inf, -3.4e38, -np.pi, -1.0, -1e-45,
                     0.0, 1e-45, 1.0, np.pi, 3.4e38, np.inf],
                    dtype=np.float32)
    keys = _f32_key(vals)
    assert (np.diff(keys.astype(np.uint64)) > 0).all()
    assert _f32_key(np.array([-0.0], np.float32))[0] < \
        _f32_key(np.array([0.0], np.float32))[0]
    # random finite floats: key order == value order
    rng = np.random.default_rng(8)
    v = (rng.standard_normal(5000) *
         10.0 ** rng.integers(-30, 30, 5000)).astype(np.float32)
    v = v[np.isfinite(v)]
    order_v = np.argsort(v, kind="stable")
    order_k = np.argsort(_f32_key(v).astype(np.uint64), kind="stable")
    # compare by value sequence (equal values may permute, none here)
    assert np.array_equal(v[order_v], v[order_k])
