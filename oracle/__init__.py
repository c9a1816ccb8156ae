# oracle — CPU restatement of the Gamma hot path. TEST INFRASTRUCTURE ONLY:
# importable from tests/, __graft_entry__.smoke() and bench.py's
# cpu_baseline leg. The product path (libgamma.so, HIP) never imports this.
from .gamma_oracle import (  # noqa: F401
    RefLib,
    gen_clustered,
    gen_queries,
    flat_topk_f64,
    flat_search,
    kmeans,
    pq_train,
    OracleIVFPQ,
    recall_at,
)
