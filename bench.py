#!/usr/bin/env python3
"""bench.py — BASELINE.json's headline metric on the MI355X engine.

Default (no flags): the north-star workload on 1 GPU —
IVFPQ d=128, m=32, nbits=8, N=10M, nlist=4096, nprobe=32, k=10 with
exact re-rank (recall_num=200), synthetic clustered float32 (seed 42),
queries = perturbed DB samples. One "step" = one batched search of nq
queries with the queries already resident in HBM (GammaCacheQueries).

Multi-GPU (launched by the driver via torch.distributed.run): one engine
(= one Vearch partition) per rank/GPU; the N-vector table is sharded
round-robin over the partitions exactly as Vearch splits a space
(STRONG scaling — the metric "N=10M ... 1/2/4/8 MI355X" holds N fixed,
and BASELINE's >=6x-at-8-GPUs target is aggregate QPS on the same
table). Per step every rank searches its shard and ONE RCCL all-gather
over xGMI moves the per-partition top-k (packed (dist,id) keys) to
rank 0, which merges with the router's semantics (client.go:1497/1558)
by one sort over world*k keys.

Prints ONE JSON line from rank 0 (driver contract), including the
roofline of the dominant kernel (the ADC list scan) and the CPU-oracle
baseline (kind "port") timed on this host.
"""
import argparse
import json
import os
import sys
import time

# pin this rank's GPU before torch / libgamma load anything
_local_rank = int(os.environ.get("LOCAL_RANK", "0"))
if "WORLD_SIZE" in os.environ and int(os.environ["WORLD_SIZE"]) > 1:
    os.environ.setdefault("HIP_VISIBLE_DEVICES", str(_local_rank))

import numpy as np  # noqa: E402

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

WORKLOADS = {
    # the config BASELINE.json's metric is quoted on (configs[3]).
    # nlist=16384 after the round-2 sweep (profiles/r02_nlist_sweep.json,
    # the reference's own ncentroids sizing sweeps,
    # internal/engine/benchs/README.md:33-62): at the pinned nprobe=32
    # it holds recall@10 = 1.0 against engine-independent fp64 ground
    # truth and cuts the 3.5x list-skew scan waste to 1.17x.
    # training_threshold follows the reference's nlist*39 rule
    # (ivfpq.cc:142).
    "ivfpq_d128_n10m_nprobe32": dict(
        kind="IVFPQ", d=128, n=10_000_000, nlist=16384, m=32, nprobe=32,
        nq=10_000, k=10, rerank=200, train_n=640_000),
    # round-1 headline shape (nlist=4096) kept as a secondary line
    "ivfpq_d128_n10m_nlist4096": dict(
        kind="IVFPQ", d=128, n=10_000_000, nlist=4096, m=32, nprobe=32,
        nq=10_000, k=10, rerank=200, train_n=160_000),
    # the reference's default nlist=2048 (ivfpq.cc:112 — §8d
    # asks for this secondary report; same 10M docs, coarser lists)
    "ivfpq_d128_n10m_nlist2048": dict(
        kind="IVFPQ", d=128, n=10_000_000, nlist=2048, m=32, nprobe=32,
        nq=10_000, k=10, rerank=200, train_n=160_000),
    # configs[1]: FLAT d=128 N=1M nq=10k (parity/secondary line)
    "flat_d128_n1m": dict(
        kind="FLAT", d=128, n=1_000_000, nlist=0, m=0, nprobe=0,
        nq=10_000, k=10, rerank=0, train_n=0),
    # configs[2]: IVFFLAT d=128 N=10M nlist=4096 nprobe=32
    "ivfflat_d128_n10m": dict(
        kind="IVFFLAT", d=128, n=10_000_000, nlist=4096, m=0, nprobe=32,
        nq=10_000, k=10, rerank=0, train_n=160_000),
    # configs[4] shape (d=768, m=96) at single-GPU scale; the 8-GPU
    # N=100M version is this sharded 8 ways by the driver's SCALE run
    "ivfpq_d768_n2m": dict(
        kind="IVFPQ", d=768, n=2_000_000, nlist=4096, m=96, nprobe=32,
        nq=2_000, k=10, rerank=200, train_n=160_000),
}

HBM_PEAK_GBS = 8000.0  # MI355X HBM3E spec peak (MI355X_MICROARCH.md)

# Measured HBM bytes per scan-kernel launch (rocprofv3 --pmc FETCH_SIZE,
# counters-only pass per the MI355X guide; raw KB x2 gfx950 wide-read
# correction). Collected on the exact workload shape keyed here; see
# profiles/r01_fetch_size_c1.csv and profiles/README.md. Any other
# shape reports traffic=null rather than guessing.
MEASURED_TRAFFIC_BYTES = {
    # r02 S-term scan at nlist=16384: rocprofv3 --pmc FETCH_SIZE
    # 4.5613 GB raw/dispatch x2 gfx950 wide-read correction = 9.12 GB
    # true per 10k-query step (profiles/r02_fetch_size.csv) — equal to
    # the 9.11 GB of algorithmic bytes actually scanned (22.8k
    # codes/query x 40 B), i.e. zero table/re-read waste. r01 at
    # nlist=4096 with per-list B staging was 112.2 GB
    # (profiles/r01_fetch_size_c1.csv).
    "ivfpq_d128_n10m_nprobe32": 9.12e9,
}


def log(msg):
    if int(os.environ.get("RANK", "0")) == 0:
        print(f"[bench] {msg}", file=sys.stderr, flush=True)


def gen_data(n, d, seed, ncl=10000, sigma=0.10, chunk=1_000_000):
    """Clustered-Gaussian DB (BASELINE.md protocol), chunked to bound RAM."""
    rng = np.random.default_rng(seed)
    centers = rng.random((ncl, d), dtype=np.float32)
    out = np.empty((n, d), dtype=np.float32)
    for s in range(0, n, chunk):
        e = min(s + chunk, n)
        asg = rng.integers(0, ncl, size=e - s)
        out[s:e] = centers[asg]
        out[s:e] += sigma * rng.standard_normal((e - s, d),
                                                dtype=np.float32)
    return out


def build_engine(cfg, base, rank, path=None):
    from vearch_amd import GammaEngine
    eng = GammaEngine(path=path or f"/tmp/gamma_bench_r{rank}")
    if cfg["kind"] == "FLAT":
        params = '{"metric_type": "L2"}'
    else:
        params = (
            '{"ncentroids": %d, %s"metric_type": "L2", '
            '"training_threshold": %d, "bucket_max_size": 12800000}'
            % (cfg["nlist"],
               f'"nsubvector": {cfg["m"]}, ' if cfg["kind"] == "IVFPQ"
               else "",
               cfg["train_n"]))
    eng.create_table(cfg["d"], cfg["kind"], params)
    t0 = time.time()
    step = 1_000_000
    for s in range(0, base.shape[0], step):
        eng.add(base[s:s + step])
    log(f"rank{rank}: added {base.shape[0]} vecs in {time.time()-t0:.1f}s")
    t0 = time.time()
    eng.build_index()
    log(f"rank{rank}: build_index in {time.time()-t0:.1f}s")
    return eng


def compute_recall(eng, cfg, queries, nq_gt=512):
    """recall@k vs exact brute-force over THIS partition (engine FLAT
    path, bit-exact — parity-tested against the oracle)."""
    q = queries[:nq_gt]
    gd, gi = eng.raw_search(q, cfg["k"], nprobe=cfg["nprobe"],
                            rerank=cfg["rerank"])
    from vearch_amd import proto  # noqa: F401
    res = eng.search_pb(q, topn=100, brute=1)
    hits = 0
    for t in range(len(q)):
        gt = {int(it["fields"]["_id"]) for it in res[t]["items"][:cfg["k"]]}
        got = set(int(x) for x in gi[t] if x >= 0)
        hits += len(gt & got)
    return hits / (len(q) * cfg["k"])


def cpu_baseline(eng, cfg, queries, budget_s=20.0):
    """The CPU oracle (oracle/ref_scan.c, OpenMP 'port') on the same
    trained model + lists, on a bounded query sample."""
    import oracle as orc
    from oracle.gamma_oracle import RefLib
    if cfg["kind"] != "IVFPQ":
        return None
    d, nlist, M = cfg["d"], cfg["nlist"], cfg["m"]
    ox = orc.OracleIVFPQ(d, nlist, M, metric="L2")
    cent, books = eng.debug_model(nlist, d, M)
    ox.centroids, ox.codebooks = cent, books
    ids_all, codes_all, offsets = [], [], [0]
    for ln in range(nlist):
        li, lc = eng.debug_list(ln, M)
        ids_all.append(li)
        codes_all.append(lc)
        offsets.append(offsets[-1] + len(li))
    ox.ids = np.concatenate(ids_all)
    ox.codes = (np.concatenate(codes_all) if ids_all else
                np.empty((0, M), np.uint8))
    ox.offsets = np.array(offsets, dtype=np.int64)
    cores = RefLib.lib().oracle_num_threads()
    # calibrate sample size: run 32 queries, scale to ~budget_s
    sample = 32
    t0 = time.time()
    ox.search(queries[:sample], cfg["k"], cfg["nprobe"])
    dt = time.time() - t0
    per_q = dt / sample
    sample = int(max(32, min(len(queries), budget_s / max(per_q, 1e-6))))
    t0 = time.time()
    ox.search(queries[:sample], cfg["k"], cfg["nprobe"])
    dt = time.time() - t0
    return {
        "value": round(sample / dt, 3),
        "unit": "QPS",
        "cores": cores,
        "kind": "port",
        "sample": f"{sample} queries of the same workload "
                  f"(~{dt:.1f}s CPU, OpenMP over {cores} cores)",
    }


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=5)
    ap.add_argument("--warmup", type=int, default=2)
    ap.add_argument("--workload", default="ivfpq_d128_n10m_nprobe32",
                    choices=sorted(WORKLOADS))
    ap.add_argument("--db-size", type=int, default=0,
                    help="override DB size (N per GPU)")
    ap.add_argument("--queries", type=int, default=0,
                    help="override nq per step")
    ap.add_argument("--nprobe", type=int, default=0)
    ap.add_argument("--nlist", type=int, default=0,
                    help="override ncentroids (training_threshold scales "
                         "with it, reference ivfpq.cc:142 nlist*39 rule)")
    ap.add_argument("--rerank", type=int, default=-1,
                    help="recall_num (-1 = workload default)")
    ap.add_argument("--skip-recall", action="store_true")
    ap.add_argument("--skip-cpu-baseline", action="store_true")
    ap.add_argument("--index-dir", default="",
                    help="reuse a built index: load from DIR if a dump "
                         "exists there, else build and dump to DIR")
    ap.add_argument("--prepare-only", action="store_true",
                    help="with --index-dir: build+dump, skip the bench")
    args = ap.parse_args()

    cfg = dict(WORKLOADS[args.workload])
    if args.db_size:
        cfg["n"] = args.db_size
        cfg["train_n"] = min(cfg["train_n"], cfg["n"])
    if args.queries:
        cfg["nq"] = args.queries
    if args.nprobe:
        cfg["nprobe"] = args.nprobe
    if args.nlist and cfg["nlist"]:
        cfg["nlist"] = args.nlist
        cfg["train_n"] = min(cfg["n"], max(cfg["train_n"],
                                           39 * args.nlist))
    if args.rerank >= 0:
        cfg["rerank"] = args.rerank

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    backend = os.environ.get("GAMMA_BENCH_BACKEND", "nccl")
    dist = None
    if world > 1:
        import torch
        import torch.distributed as tdist
        tdist.init_process_group(backend)
        if backend == "nccl":
            torch.cuda.set_device(0)  # HIP_VISIBLE_DEVICES pins the GPU
        dist = tdist

    import torch

    idir = args.index_dir
    if idir and world > 1:
        idir = os.path.join(idir, f"r{rank}")
    have_dump = bool(idir) and os.path.exists(
        os.path.join(idir, "gamma.dump")) and os.path.exists(
        os.path.join(idir, "queries.npy"))

    if have_dump:
        from vearch_amd import GammaEngine
        t0 = time.time()
        eng = GammaEngine(path=idir)
        if cfg["kind"] == "FLAT":
            params = '{"metric_type": "L2"}'
        else:
            params = (
                '{"ncentroids": %d, %s"metric_type": "L2", '
                '"training_threshold": %d, "bucket_max_size": 12800000}'
                % (cfg["nlist"],
                   f'"nsubvector": {cfg["m"]}, ' if cfg["kind"] == "IVFPQ"
                   else "", cfg["train_n"]))
        eng.create_table(cfg["d"], cfg["kind"], params)
        eng.load()
        queries = np.load(os.path.join(idir, "queries.npy"))
        cfg["nq"] = queries.shape[0]
        log(f"loaded index ({eng.num_docs()} docs) + queries in "
            f"{time.time()-t0:.1f}s")
    else:
        # one table of N vectors, sharded round-robin over partitions
        # (shard-local docid * world + rank = global docid); every rank
        # derives the identical query set from the same seeds
        t0 = time.time()
        base_full = gen_data(cfg["n"], cfg["d"], seed=42)
        rng = np.random.default_rng(43)
        idx = rng.integers(0, cfg["n"], size=cfg["nq"])
        queries = base_full[idx] + 0.05 * rng.standard_normal(
            (cfg["nq"], cfg["d"]), dtype=np.float32)
        base = np.ascontiguousarray(base_full[rank::world])             if world > 1 else base_full
        del base_full
        log(f"data gen in {time.time()-t0:.1f}s "
            f"(shard {base.shape[0]} of {cfg['n']})")
        if idir:
            os.makedirs(idir, exist_ok=True)
        eng = build_engine(cfg, base, rank, path=idir or None)
        if idir:
            eng.dump()
            np.save(os.path.join(idir, "queries.npy"), queries)
            log("index dumped")
            if args.prepare_only:
                eng.close()
                return
    nq = eng.cache_queries(queries)

    # actual scanned codes/query (clustered data probes the dense lists,
    # so this exceeds nprobe*N/nlist; the roofline stays on algorithmic
    # bytes per BASELINE.md)
    eff_codes = None
    if rank == 0 and cfg["kind"] in ("IVFPQ", "IVFFLAT"):
        try:
            from vearch_amd.engine import lib as _lib
            _, pl = eng.debug_coarse_assign(queries[:256], cfg["nprobe"])
            sizes = {int(ln): _lib().GammaDebugGetList(
                eng.h, int(ln), None, None)
                for ln in np.unique(pl) if ln >= 0}
            eff_codes = float(np.mean(
                [sum(sizes.get(int(x), 0) for x in row) for row in pl]))
        except Exception:
            pass

    recall = None
    if rank == 0 and not args.skip_recall and world == 1:
        t0 = time.time()
        recall = compute_recall(eng, cfg, queries)
        log(f"recall@{cfg['k']} = {recall:.4f} ({time.time()-t0:.1f}s)")

    def step():
        return eng.search_cached(nq, cfg["k"], nprobe=cfg["nprobe"],
                                 rerank=cfg["rerank"])

    from vearch_amd.merge import pack_keys_signed_torch, unpack_keys_signed

    def gather_and_merge(dists, ids):
        if world == 1:
            return dists, ids
        # globalize round-robin shard ids + pack (dist,id) into signed
        # int64 keys on the GPU, ONE all-gather over xGMI (~nq*k*8 B per
        # rank), merge = one torch sort over world*k keys
        # (client.go:1497 semantics; parity-tested vs merge_topk)
        d_t = torch.from_numpy(dists)
        i_t = torch.from_numpy(ids)
        if backend == "nccl":
            d_t, i_t = d_t.cuda(), i_t.cuda()
        keys = pack_keys_signed_torch(d_t, i_t, world, rank)
        out = [torch.empty_like(keys) for _ in range(world)]
        dist.all_gather(out, keys)
        if rank == 0:
            merged = torch.sort(torch.cat(out, dim=1),
                                dim=1).values[:, :cfg["k"]]
            return unpack_keys_signed(merged.cpu().numpy())
        return None, None

    # warmup
    for _ in range(args.warmup):
        gather_and_merge(*step())

    scan_ms = []
    if world > 1:
        dist.barrier()
    torch.cuda.synchronize()
    t_start = time.time()
    for _ in range(args.steps):
        gather_and_merge(*step())
        scan_ms.append(eng.last_timing()["scan_us"] / 1000.0)
    torch.cuda.synchronize()
    if world > 1:
        dist.barrier()
    elapsed = time.time() - t_start
    if world > 1:
        el = torch.tensor([elapsed])
        if backend == "nccl":
            el = el.cuda()
        dist.all_reduce(el, op=dist.ReduceOp.MAX)
        elapsed = float(el.item())

    if rank == 0:
        qps = cfg["nq"] * args.steps / elapsed
        timing = eng.last_timing()
        # roofline of the dominant kernel (ADC list scan / flat scan)
        if cfg["kind"] == "IVFPQ":
            per_q_bytes = cfg["nprobe"] * (cfg["n"] / cfg["nlist"]) * \
                (cfg["m"] + 8)
        elif cfg["kind"] == "IVFFLAT":
            per_q_bytes = cfg["nprobe"] * (cfg["n"] / cfg["nlist"]) * \
                cfg["d"] * 4
        else:
            per_q_bytes = cfg["n"] * cfg["d"] * 4
        t_scan_s = float(np.mean(scan_ms)) / 1000.0 if scan_ms else None
        roofline = None
        if t_scan_s and t_scan_s > 0:
            achieved = cfg["nq"] * per_q_bytes / t_scan_s / 1e9
            roofline = {
                "bound": "hbm",
                "achieved": round(achieved, 1),
                "peak": HBM_PEAK_GBS,
                "unit": "GB/s",
                "frac": round(achieved / HBM_PEAK_GBS, 4),
                "traffic": MEASURED_TRAFFIC_BYTES.get(args.workload)
                if cfg["nq"] == WORKLOADS.get(args.workload, {}).get("nq")
                and not args.db_size and not args.nlist
                and not args.nprobe else None,
            }
        cpu = None
        if not args.skip_cpu_baseline and world == 1:
            try:
                cpu = cpu_baseline(eng, cfg, queries)
            except Exception as ex:  # noqa: BLE001
                log(f"cpu_baseline failed: {ex}")
        line = {
            "metric": "QPS @ recall@10, IVFPQ d=128 N=10M nprobe=32",
            "value": round(qps, 2),
            "unit": "QPS",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000.0, 3),
            "higher_is_better": True,
            "scaling": "strong",
            "vs_baseline": None,
            "dtype": "f32",
            "data": "synthetic",
            "config": {
                "workload": args.workload,
                "kind": cfg["kind"],
                "d": cfg["d"],
                "n_total": cfg["n"],
                "n_per_gpu": cfg["n"] // world,
                "nlist": cfg["nlist"],
                "m": cfg["m"],
                "nprobe": cfg["nprobe"],
                "nq_per_step": cfg["nq"],
                "topk": cfg["k"],
                "recall_num": cfg["rerank"],
                "recall_at_10": recall,
                "scanned_codes_per_query": eff_codes,
                "algorithmic_codes_per_query":
                    cfg["nprobe"] * cfg["n"] / cfg["nlist"]
                    if cfg["nlist"] else None,
                "parallelism": f"dp{world} (1 partition/GPU, RCCL "
                               "all-gather top-k merge)",
            },
            "roofline": roofline,
            "cpu_baseline": cpu,
            "stage_us_last_step": timing,
        }
        print(json.dumps(line), flush=True)

    eng.close()
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
