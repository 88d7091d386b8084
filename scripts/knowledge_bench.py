#!/usr/bin/env python3
"""Knowledge-search QPS benchmark (BASELINE config 2): GPU BGE encoder +
HIP brute-force top-k cosine over an HBM-resident corpus, vs the CPU
reference path. Writes gpurun_out/knowledge_bench.json.
"""
from __future__ import annotations

import json
import os
import random
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

WORDS = ("redis pool timeout latency gateway upstream deploy oom memory disk cpu "
         "throttle alarm error connection exhausted checkout cart payment user "
         "service cluster node pod replica scale rollback postgres shard queue").split()


def synth_corpus(n: int, seed: int = 1) -> list[str]:
    rng = random.Random(seed)
    return [" ".join(rng.choices(WORDS, k=rng.randint(8, 40))) for _ in range(n)]


def main() -> None:
    assert torch.cuda.is_available()
    from runbookai_amd import ops
    from runbookai_amd.embedding.encoder import BgeEncoder

    report = {}
    enc = BgeEncoder(device="cuda:0")

    # ---- embedding throughput ----
    docs = synth_corpus(2000)
    t0 = time.time()
    vecs = enc.encode(docs, batch_size=128)
    torch.cuda.synchronize()
    embed_s = time.time() - t0
    report["embed_docs_per_s"] = round(len(docs) / embed_s, 1)

    # ---- search QPS at several corpus sizes ----
    queries = synth_corpus(256, seed=9)
    qvecs = enc.encode(queries, batch_size=128)
    for n in (10_000, 100_000, 1_000_000):
        rng = np.random.default_rng(3)
        corpus = rng.standard_normal((n, 384), dtype=np.float32)
        corpus /= np.linalg.norm(corpus, axis=1, keepdims=True)
        matrix = torch.from_numpy(corpus).half().to("cuda:0")
        qt = torch.from_numpy(qvecs).half().to("cuda:0")
        # warmup
        ops.topk_cosine(matrix, qt[0], 5)
        torch.cuda.synchronize()
        t0 = time.time()
        iters = 200
        for i in range(iters):
            ops.topk_cosine(matrix, qt[i % len(qt)], 5)
        torch.cuda.synchronize()
        dt = time.time() - t0
        report[f"search_qps_n{n}"] = round(iters / dt, 1)
        report[f"search_ms_n{n}"] = round(dt / iters * 1000, 3)
        del matrix

    # ---- top-k parity vs CPU reference on real embedded corpus ----
    matrix = torch.from_numpy(vecs).half().to("cuda:0")
    q = torch.from_numpy(qvecs[0]).half().to("cuda:0")
    g_vals, g_idx = ops.topk_cosine(matrix, q, 5)
    from runbookai_amd.ops.reference import topk_cosine as ref_topk

    c_vals, c_idx = ref_topk(torch.from_numpy(vecs), torch.from_numpy(qvecs[0]), 5)
    report["topk_parity"] = bool(set(g_idx.cpu().tolist()) == set(c_idx.tolist()))

    # ---- end-to-end pipeline QPS (embed query + search, corpus 100k) ----
    rng = np.random.default_rng(4)
    corpus = rng.standard_normal((100_000, 384), dtype=np.float32)
    corpus /= np.linalg.norm(corpus, axis=1, keepdims=True)
    matrix = torch.from_numpy(corpus).half().to("cuda:0")
    torch.cuda.synchronize()
    t0 = time.time()
    iters = 100
    for i in range(iters):
        qv = enc.encode([queries[i % len(queries)]])[0]
        qt1 = torch.from_numpy(qv).half().to("cuda:0")
        ops.topk_cosine(matrix, qt1, 5)
    torch.cuda.synchronize()
    report["e2e_search_qps_n100k"] = round(iters / (time.time() - t0), 1)

    # batched e2e: one encoder forward per BATCH of queries (the serving
    # shape — concurrent investigations batch their lookups)
    for B in (16, 64):
        batch = [queries[i % len(queries)] for i in range(B)]
        torch.cuda.synchronize()
        t0 = time.time()
        reps = max(1, 512 // B)
        for _ in range(reps):
            qv = enc.encode(batch, batch_size=B)
            qt = torch.from_numpy(qv).half().to("cuda:0")
            scores = qt @ matrix.t()
            torch.topk(scores.float(), 5, dim=1)
        torch.cuda.synchronize()
        report[f"e2e_search_qps_n100k_batch{B}"] = round(
            reps * B / (time.time() - t0), 1)

    os.makedirs("gpurun_out", exist_ok=True)
    with open("gpurun_out/knowledge_bench.json", "w") as f:
        json.dump(report, f, indent=1)
    print(json.dumps(report, indent=1))


if __name__ == "__main__":
    main()
