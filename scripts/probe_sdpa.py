"""Probe which SDPA backend serves the BERT attention shape on MI355X."""
import time

import torch
import torch.nn.functional as F
from torch.nn.attention import SDPBackend, sdpa_kernel

q = torch.randn(64, 12, 512, 64, device="cuda", dtype=torch.bfloat16)
k, v = torch.randn_like(q), torch.randn_like(q)

for name, be in [("FLASH", SDPBackend.FLASH_ATTENTION),
                 ("EFFICIENT", SDPBackend.EFFICIENT_ATTENTION),
                 ("MATH", SDPBackend.MATH)]:
    try:
        with sdpa_kernel(be):
            o = F.scaled_dot_product_attention(q, k, v)
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(20):
                o = F.scaled_dot_product_attention(q, k, v)
            torch.cuda.synchronize()
            ms = (time.perf_counter() - t0) / 20 * 1000
        print("%s: %.3f ms" % (name, ms), flush=True)
    except Exception as e:
        print("%s: unavailable (%s)" % (name, str(e)[:120]), flush=True)

# default dispatch
o = F.scaled_dot_product_attention(q, k, v)
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(20):
    o = F.scaled_dot_product_attention(q, k, v)
torch.cuda.synchronize()
print("DEFAULT: %.3f ms" % ((time.perf_counter() - t0) / 20 * 1000))
