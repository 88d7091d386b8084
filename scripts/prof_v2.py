import torch, time, sys
sys.path.insert(0, "/root/repo")
from runbookai_amd import ops as rops
from runbookai_amd.ops import _get_ext
ext = _get_ext()
B_, Hq_, Hk_, D_, S_ = 16, 32, 8, 128, 2048
T_ = B_ * S_
q = torch.randn(T_, Hq_, D_, dtype=torch.bfloat16, device="cuda") * 0.5
k = torch.randn(T_, Hk_, D_, dtype=torch.bfloat16, device="cuda") * 0.5
v = torch.randn(T_, Hk_, D_, dtype=torch.bfloat16, device="cuda") * 0.5
starts_h = torch.arange(0, T_ + 1, S_, dtype=torch.int32)
starts = starts_h.to("cuda")
tb, tq = [t.to("cuda") for t in rops._build_qtiles(starts_h, 256)]
scale = 1.0 / (D_ ** 0.5)
for _ in range(3):
    ext.flash_prefill2(q, k, v, tb, tq, starts, scale, True)
torch.cuda.synchronize()
t0 = time.time()
for _ in range(10):
    ext.flash_prefill2(q, k, v, tb, tq, starts, scale, True)
torch.cuda.synchronize()
dt = (time.time() - t0) / 10
flops = 2 * 2 * (S_ * S_ / 2) * D_ * Hq_ * B_
print(f"v2: {dt*1000:.2f} ms  {flops/dt/1e12:.0f} TF")
