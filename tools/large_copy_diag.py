#!/usr/bin/env python3
"""Diagnose the large-buffer copy mismatch: sweep sizes across the
n4=2^32 float4-count boundary (64 GiB) and report the first mismatching
byte offset for each size."""
import os
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

import torch  # noqa: E402

from cea_amd.ops import native  # noqa: E402


def first_mismatch(a, b, chunk=1 << 26):
    n = a.numel()
    for off in range(0, n, chunk):
        ca, cb = a[off:off + chunk], b[off:off + chunk]
        if not torch.equal(ca, cb):
            ne = (ca != cb).nonzero()
            return off + ne[0].item(), int((ca != cb).sum())
    return None, 0


def main():
    native.assert_native_available()
    free = native.device_probe(0)["free_bytes"]
    print("free GiB:", free / 2**30)
    for gib in (32, 60, 63, 64, 65, 80, 120):
        if 2 * gib * 2**30 > free - (4 << 30):
            print(f"{gib} GiB: skip (not enough free)")
            continue
        n = (gib * 2**30) // 4
        a = torch.empty(n, dtype=torch.float32, device="cuda")
        # pattern: index-derived values at head/middle/tail windows
        for off in (0, n // 2 - 512, n - 1024):
            a[off:off + 1024] = torch.arange(1024, dtype=torch.float32,
                                             device="cuda") + off % 9973
        b = torch.zeros_like(a)
        native.copy_(b, a)
        torch.cuda.synchronize()
        # compare the three windows + a coarse full scan on mismatch
        bad = []
        for off in (0, n // 2 - 512, n - 1024):
            if not torch.equal(a[off:off + 1024], b[off:off + 1024]):
                bad.append(off)
        if bad:
            off, cnt = first_mismatch(a, b)
            print(f"{gib} GiB (n4={n//4}): MISMATCH windows {bad}; first bad "
                  f"elem {off} (float idx, = {off*4/2**30:.3f} GiB), "
                  f"~{cnt} bad in that chunk")
        else:
            print(f"{gib} GiB (n4={n//4}): ok")
        del a, b
        torch.cuda.empty_cache()


if __name__ == "__main__":
    main()
