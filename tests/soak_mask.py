#!/usr/bin/env python3
"""GPU-vs-oracle mask/verify parity soak (not a pytest; run via gpurun):

    python3 tests/soak_mask.py [seconds]

Random committees across the dispatch regimes (tiny / compacted <2048 /
windowed >=2048 incl. ragged n), random masks at several densities plus
adversarial negation pairs, and batched aggregate-verifies.  Exits nonzero
on the first mismatch."""
import os
import random
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from harmony_amd import core          # noqa: E402
from oracle import capi               # noqa: E402
from oracle import pyref as pr        # noqa: E402


def sk_bytes(i):
    return pr.fr_serialize(pr.synth_sk(i))


def main():
    budget = float(sys.argv[1]) if len(sys.argv) > 1 else 120.0
    t0 = time.time()
    rng = random.Random(20260915)
    sizes = [17, 100, 1024, 2048, 2053, 4096, 8191, 16384]
    trials = 0
    rounds = 0
    while time.time() - t0 < budget:
        n = sizes[rounds % len(sizes)]
        rounds += 1
        base = rng.randrange(1 << 20)
        sks = b"".join(sk_bytes(base + i) for i in range(n))
        pks = bytearray(core.batch_pk_from_sk(sks, n))
        if n >= 16 and rng.random() < 0.5:
            # negation pair inside one 8-key window group
            g = rng.randrange(1, n // 8)
            pks[(8 * g + 1) * 48:(8 * g + 2) * 48] = pks[8 * g * 48:(8 * g + 1) * 48]
            pks[(8 * g + 2) * 48 - 1] ^= 0x80
        pks = bytes(pks)
        gc = core.Committee(pks, n)
        oc = capi.Committee(pks, n)
        nb = (n + 7) // 8
        for density in (0.03, 0.33, 0.9, 1.0):
            bm = bytearray(nb)
            for i in range(n):
                if rng.random() < density:
                    bm[i >> 3] |= 1 << (i & 7)
            bm = bytes(bm)
            g = gc.mask_aggregate(bm)
            o = oc.mask_aggregate(bm)
            if g != o:
                print(f"MISMATCH n={n} density={density} bm[:8]={bm[:8].hex()}")
                print(" gpu:", g.hex())
                print(" ora:", o.hex())
                return 1
            trials += 1
    print(f"soak ok: {trials} mask comparisons over {rounds} committees "
          f"in {time.time() - t0:.0f}s")
    return 0


if __name__ == "__main__":
    sys.exit(main())
