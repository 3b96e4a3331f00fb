"""AdamW devstep kernel micro-benchmark (width A/B via
MODALITIES_AMD_ADAMW_WIDE)."""
import os
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch

from modalities_amd.ops.backend import hip_ext

n = 320_000_000
p0 = torch.randn(n, device="cuda")
g0 = torch.randn(n, device="cuda")
m0 = torch.zeros(n, device="cuda")
v0 = torch.zeros(n, device="cuda")
mask = torch.ones(n, device="cuda")
step = torch.tensor(3, dtype=torch.int32, device="cuda")
out = torch.empty(n, device="cuda", dtype=torch.bfloat16)

def call():
    hip_ext().fused_adamw_masked_devstep(p0, g0, m0, v0, mask, step, out,
                                         None, 3e-4, .9, .95, 1e-8, .1)

for _ in range(3):
    call()
torch.cuda.synchronize()
s, e = torch.cuda.Event(True), torch.cuda.Event(True)
s.record()
for _ in range(20):
    call()
e.record()
torch.cuda.synchronize()
ms = s.elapsed_time(e) / 20
print(f"wide={os.environ.get('MODALITIES_AMD_ADAMW_WIDE', '2(default)')}: "
      f"{ms:.3f} ms  ({n * 34 / 1e9 / (ms / 1e3):.0f} GB/s)")
