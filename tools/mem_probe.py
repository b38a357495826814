#!/usr/bin/env python3
"""Peak-memory probe: 30 full V3-Large B256 steps, report allocator peaks
(leak check: peak measured after warmup must be flat)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from yet_another_mobilenet_series_amd import models
from yet_another_mobilenet_series_amd.ops import functional as OF
from yet_another_mobilenet_series_amd.ops.backend import load_extension
from yet_another_mobilenet_series_amd.utils.optim import FusedRMSpropTF


def main():
    load_extension(required=True)
    m = (models.build_model("mobilenet_v3_large", dropout=0.2).cuda()
         .to(memory_format=torch.channels_last).train())
    opt = FusedRMSpropTF(m.parameters(), lr=0.01)
    x = (torch.randn(256, 3, 224, 224, device="cuda").bfloat16()
         .contiguous(memory_format=torch.channels_last))
    y = torch.randint(0, 1000, (256,), device="cuda")
    peaks = []
    for i in range(30):
        opt.zero_grad(set_to_none=True)
        loss = OF.label_smooth_ce(m(x), y, 0.1)
        loss.backward()
        opt.step()
        if i >= 4:
            torch.cuda.synchronize()
            peaks.append(torch.cuda.max_memory_allocated())
            torch.cuda.reset_peak_memory_stats()
    gib = [round(p / 2**30, 3) for p in peaks]
    print("per-step peak GiB (post-warmup):", gib[:3], "...", gib[-3:])
    print("reserved GiB:", round(torch.cuda.max_memory_reserved() / 2**30,
                                 2))
    assert max(gib[5:]) <= gib[2] + 0.01, "allocator peak grew (leak?)"
    print("flat peak OK")


if __name__ == "__main__":
    main()
