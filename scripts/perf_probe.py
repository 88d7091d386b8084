#!/usr/bin/env python3
"""Engine performance probe (GPU): per-step decode latency at several batch
sizes (graphed vs eager) and prefill throughput. Writes JSON to
gpurun_out/perf_probe.json for tracking.

Usage: python scripts/perf_probe.py [--model llama3-8b] [--quick]
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def bench_decode(model, kv, B: int, ctx: int, iters: int, use_graphs: bool) -> float:
    """Returns ms/step for a decode step with B sequences at context ctx."""
    model.use_graphs = use_graphs
    seq_ids = list(range(1000, 1000 + B))
    for s in seq_ids:
        kv.allocate(s, ctx + iters + 8)
        kv.set_len(s, ctx)
    try:
        bt, lens = kv.batch_tables(seq_ids, model.device)
        ids = torch.randint(0, 255, (B,), dtype=torch.long)
        pos = torch.full((B,), ctx - 1, dtype=torch.int32)
        slots = torch.cat([kv.slot_mapping(s, ctx - 1, 1) for s in seq_ids])
        # warmup
        for _ in range(3):
            model.decode(ids, pos, bt, lens, slots)
        torch.cuda.synchronize()
        t0 = time.time()
        for _ in range(iters):
            model.decode(ids, pos, bt, lens, slots)
        torch.cuda.synchronize()
        return (time.time() - t0) / iters * 1000.0
    finally:
        for s in seq_ids:
            kv.free(s)


def bench_prefill(model, kv, T: int, iters: int) -> float:
    """Returns tokens/s for a single-sequence prefill of T tokens."""
    times = []
    for it in range(iters + 1):
        sid = 5000 + it
        kv.allocate(sid, T)
        ids = torch.randint(0, 255, (T,), dtype=torch.long)
        pos = torch.arange(T, dtype=torch.int32)
        starts = torch.tensor([0, T], dtype=torch.int32)
        slots = kv.slot_mapping(sid, 0, T)
        torch.cuda.synchronize()
        t0 = time.time()
        model.prefill(ids, pos, starts, slots)
        torch.cuda.synchronize()
        times.append(time.time() - t0)
        kv.free(sid)
    times = times[1:]  # drop warmup
    return T / (sum(times) / len(times))


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="llama3-8b")
    p.add_argument("--quick", action="store_true")
    args = p.parse_args()
    assert torch.cuda.is_available()
    from runbookai_amd.engine.llama import CONFIGS, LlamaModel

    model = LlamaModel(CONFIGS[args.model], device="cuda:0", kv_blocks=4096)
    kv = model.kv
    report: dict = {"model": args.model, "device": torch.cuda.get_device_name(0)}

    iters = 10 if args.quick else 30
    decode = {}
    for B in ([1, 8] if args.quick else [1, 2, 4, 8, 16, 32]):
        for mode in ("graph", "eager"):
            ms = bench_decode(model, kv, B, ctx=512, iters=iters, use_graphs=mode == "graph")
            decode[f"B{B}_{mode}_ms"] = round(ms, 3)
            decode[f"B{B}_{mode}_tok_s"] = round(B / ms * 1000.0, 1)
    report["decode_ctx512"] = decode

    # context-length sweep at B=8 (graphed)
    ctx_sweep = {}
    for ctx in ([512] if args.quick else [128, 512, 2048, 4096]):
        ms = bench_decode(model, kv, 8, ctx=ctx, iters=iters, use_graphs=True)
        ctx_sweep[f"ctx{ctx}_ms"] = round(ms, 3)
    report["decode_B8_ctx_sweep"] = ctx_sweep

    prefill = {}
    for T in ([512] if args.quick else [512, 2048]):
        tps = bench_prefill(model, kv, T, iters=2 if args.quick else 4)
        prefill[f"T{T}_tok_s"] = round(tps, 1)
    report["prefill"] = prefill

    # attention kernels in isolation (Llama-3-8B shapes)
    from runbookai_amd import ops as rops

    attn = {}
    B_, Hq_, Hk_, D_, S_ = 16, 32, 8, 128, 2048
    T_ = B_ * S_
    q = torch.randn(T_, Hq_, D_, dtype=torch.bfloat16, device="cuda") * 0.5
    k = torch.randn(T_, Hk_, D_, dtype=torch.bfloat16, device="cuda") * 0.5
    v = torch.randn(T_, Hk_, D_, dtype=torch.bfloat16, device="cuda") * 0.5
    starts_h = torch.arange(0, T_ + 1, S_, dtype=torch.int32)
    starts = starts_h.to("cuda")
    # causal flops: 2 matmuls * 2*S^2/2*D per (head, seq)
    flops = 2 * 2 * (S_ * S_ / 2) * D_ * Hq_ * B_
    scale = 1.0 / (D_ ** 0.5)
    from runbookai_amd.ops import _get_ext as _ge
    ext_a = _ge()
    variants = {
        "v1_16x16": (ext_a.flash_prefill, 64),
        "v2_inreg": (ext_a.flash_prefill2, 256),
    }
    # within-probe interleaved A/B (guide rule 24): 6 rounds each, report min
    times = {name: [] for name in variants}
    tiles = {name: tuple(t.to("cuda") for t in rops._build_qtiles(starts_h, qt))
             for name, (_, qt) in variants.items()}
    for name, (fn, _) in variants.items():
        fn(q, k, v, *tiles[name], starts, scale, True)   # warmup
    torch.cuda.synchronize()
    for _ in range(6):
        for name, (fn, _) in variants.items():
            torch.cuda.synchronize()
            t0 = time.time()
            for _ in range(3):
                fn(q, k, v, *tiles[name], starts, scale, True)
            torch.cuda.synchronize()
            times[name].append((time.time() - t0) / 3)
    for name, ts in times.items():
        best = min(ts)
        attn[f"flash_{name}_ms"] = round(best * 1000, 2)
        attn[f"flash_{name}_TF"] = round(flops / best / 1e12, 1)
    del q, k, v
    report["attention"] = attn

    # skinny GEMM vs hipBLASLt on the decode projection shapes
    from runbookai_amd.ops import _get_ext

    ext = _get_ext()
    gemm = {}
    for (name, N, K) in (("qkv", 6144, 4096), ("o", 4096, 4096),
                         ("gate_up", 28672, 4096), ("down", 4096, 14336),
                         ("lm_head", 128256, 4096)):
        for M in (1, 8, 32):
            x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
            w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda")
            variants = [("skinny", lambda: ext.skinny_gemm(x, w)),
                        ("blaslt", lambda: x @ w.t())]
            if M <= 4:
                variants.append(("gemv", lambda: ext.decode_gemv(
                    x, w, None, None, 0, 1e-5)))
            for label, fn in variants:
                for _ in range(3):
                    fn()
                torch.cuda.synchronize()
                t0 = time.time()
                for _ in range(20):
                    fn()
                torch.cuda.synchronize()
                us = (time.time() - t0) / 20 * 1e6
                tbps = (N * K * 2) / (us / 1e6) / 1e12
                gemm[f"{name}_M{M}_{label}"] = f"{us:.1f}us {tbps:.2f}TB/s"
            del x, w
    report["gemm_shapes"] = gemm

    os.makedirs("gpurun_out", exist_ok=True)
    with open("gpurun_out/perf_probe.json", "w") as f:
        json.dump(report, f, indent=1)
    print(json.dumps(report, indent=1))


if __name__ == "__main__":
    main()
