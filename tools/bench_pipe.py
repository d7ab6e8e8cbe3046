#!/usr/bin/env python3
"""Async input-pipe (HostPinnedPrefetcher) throughput evidence.

BASELINE config #5 names the async data pipe; bench.py's timed region
excludes H2D (disclosed), so this micro-bench supplies the perf
evidence: images/s the pipe can source vs what the b256 ResNet-152 /
b2048 ResNet-50 steps consume.  Run on a GPU box:
    python tools/bench_pipe.py
"""

import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch  # noqa: E402

from deeplearning_cfn_amd.data import HostPinnedPrefetcher, SyntheticImages


def run(batch, image_size, n_batches=30):
    # pool=4: host tensors are synthesised once and pinned once, so the
    # measurement is the PIPE (queue + H2D on the side stream), not the
    # CPU's randn rate (fresh-randn sources ~1.3k img/s of host work —
    # which is why bench.py pre-stages device batches, as disclosed)
    src = SyntheticImages(batch, image_size=image_size, seed=1,
                          dtype=torch.bfloat16, channels_last=True,
                          pinned=True, pool=4)
    pf = HostPinnedPrefetcher(src, torch.device("cuda", 0))
    it = iter(pf)
    # warm
    for _ in range(4):
        x, y = next(it)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n_batches):
        x, y = next(it)
        # consume: force the copy to be complete before the next fetch
        torch.cuda.current_stream().synchronize()
    dt = time.perf_counter() - t0
    pf.close()
    ips = batch * n_batches / dt
    gbs = batch * n_batches * 3 * image_size * image_size * 2 / dt / 1e9
    print(f"pipe b{batch} {image_size}^2 bf16: {ips:,.0f} img/s "
          f"({gbs:.1f} GB/s H2D incl host synth)")
    return ips


if __name__ == "__main__":
    run(256, 224)
    run(2048, 224)
