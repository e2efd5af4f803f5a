"""CPU tests for bench.py's input generators: the signer-key sums gate the
bench's correctness check, so their exact-arithmetic trick (8-bit chunk
float32 sgemm, sums < 2^24) must itself be pinned."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import bench  # noqa: E402


def test_signer_sums_exact():
    import numpy as np
    from oracle import pyref as pr
    rng = np.random.default_rng(77)
    n = 700
    sk_ints = [pr.synth_sk(i) for i in range(n)]
    chunks8 = np.array([[(s >> (8 * j)) & 0xFF for j in range(32)]
                        for s in sk_ints], dtype=np.float32)
    bits = bench._rand_bits(rng, 5, n, 0.5)
    sums = bench._signer_sums(bits, chunks8, pr.R)
    for b in range(5):
        want = sum(sk_ints[i] for i in np.nonzero(bits[b])[0]) % pr.R
        assert sums[b] == want


def test_rand_bits_deterministic_and_dense():
    import numpy as np
    a = bench._rand_bits(np.random.default_rng(5), 4, 1000, 0.9)
    b = bench._rand_bits(np.random.default_rng(5), 4, 1000, 0.9)
    assert (a == b).all()
    density = a.mean()
    assert 0.85 < density < 0.95
