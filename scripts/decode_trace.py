#!/usr/bin/env python3
"""Decode-step kernel breakdown at B=1 (llama3-8b): run N graphed decode
steps under rocprofv3 kernel-trace and print the per-kernel time table."""
import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from runbookai_amd.engine.llama import CONFIGS, LlamaModel

model = LlamaModel(CONFIGS["llama3-8b"], device="cuda:0", kv_blocks=2048)
kv = model.kv
B, ctx = 1, 512
for s in range(B):
    kv.allocate(100 + s, ctx + 64)
    kv.set_len(100 + s, ctx)
bt, lens = kv.batch_tables([100 + s for s in range(B)], model.device)
ids = torch.randint(0, 255, (B,), dtype=torch.long)
pos = torch.full((B,), ctx - 1, dtype=torch.int32)
slots = torch.cat([kv.slot_mapping(100 + s, ctx - 1, 1) for s in range(B)])
for _ in range(3):
    model.decode(ids, pos, bt, lens, slots)
torch.cuda.synchronize()
t0 = time.time()
N = 20
for _ in range(N):
    model.decode(ids, pos, bt, lens, slots)
torch.cuda.synchronize()
print(f"B=1 ctx=512 graphed decode: {(time.time()-t0)/N*1000:.3f} ms/step")
