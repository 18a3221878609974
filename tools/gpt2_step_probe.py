"""One GPT-2-small training step on GPU (rocprofv3 target: shows the native
layernorm/gelu/attention kernels in the kernel-stats list)."""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from hypha_amd import models

m = models.build("gpt2-small").to("cuda", torch.bfloat16)
ids = torch.randint(0, 50304, (4, 512), device="cuda")
for _ in range(3):
    loss = m(ids, labels=ids)
    loss.backward()
torch.cuda.synchronize()
print("gpt2 step ok", float(loss))
