"""Micro-benchmark harness (reference parity: infomesh/benchmarks.py —
avg/median/p95/p99/ops-sec over pipeline pieces; surfaced by
`infomesh-amd bench`)."""
from __future__ import annotations

import time
from typing import Any, Callable


def time_fn(fn: Callable[[], Any], iterations: int = 200,
            warmup: int = 10) -> dict[str, float]:
    for _ in range(warmup):
        fn()
    samples = []
    for _ in range(iterations):
        t0 = time.perf_counter()
        fn()
        samples.append((time.perf_counter() - t0) * 1e3)
    samples.sort()
    n = len(samples)
    return {
        "avg_ms": round(sum(samples) / n, 4),
        "p50_ms": round(samples[n // 2], 4),
        "p95_ms": round(samples[min(n - 1, int(n * 0.95))], 4),
        "p99_ms": round(samples[min(n - 1, int(n * 0.99))], 4),
        "ops_per_sec": round(1000.0 * n / sum(samples), 1),
    }


def run_micro_suite(iterations: int = 200) -> dict[str, dict]:
    """The CPU pipeline micro-suite (reference cli bench list)."""
    from ..search.nlp import expand_query, parse_query_filters
    from ..search.passage import classify_intent, split_passages
    from ..search.cjk import contains_cjk, tokenize_query_cjk
    from ..crawler.simhash import simhash
    from ..index.gpu_index import bm25_term_ids

    text = ("The MI355X accelerator runs HIP kernels on CDNA4 compute "
            "units with matrix cores and LDS tiling. " * 20)
    query = "gpu kernel performance site:rocm.docs.amd.com after:2024-01-01"

    return {
        "query_expansion": time_fn(lambda: expand_query("fast gpu error"),
                                   iterations),
        "nlp_filter_parse": time_fn(lambda: parse_query_filters(query),
                                    iterations),
        "passage_split": time_fn(lambda: split_passages(text), iterations),
        "cjk_detect": time_fn(lambda: contains_cjk(text), iterations),
        "cjk_tokenize": time_fn(lambda: tokenize_query_cjk("東京タワー観光"),
                                iterations),
        "intent_classify": time_fn(lambda: classify_intent(
            "how do I profile kernels"), iterations),
        "simhash": time_fn(lambda: simhash(text), max(iterations // 10, 10)),
        "bm25_tokenize": time_fn(lambda: bm25_term_ids(text),
                                 max(iterations // 10, 10)),
    }
