import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from automodel_amd.ops._backend import hip_ops
ops = hip_ops()
# linear pattern: lds element i has value i (exact in bf16 up to 256)
n = 512
pat = torch.arange(n, dtype=torch.float32, device="cuda")
out_lo = ops.tr16_probe((pat % 256).to(torch.bfloat16))          # low 8 bits
out_hi = ops.tr16_probe((pat // 256).to(torch.bfloat16))         # high bit
idx = (out_lo + 256 * out_hi).long()  # [2,64,4] element index each slot got
for rd in range(1):
    print(f"--- read {rd}: per-lane element indices (lane: [j0..j3])")
    for l in range(0, 64, 1):
        v = idx[rd, l].tolist()
        print(f"l{l:02d}: {v}", end="   ")
        if l % 4 == 3: print()
