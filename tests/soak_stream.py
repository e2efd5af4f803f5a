"""Randomized stream-context soak (run explicitly on a GPU box, not part of
the default suite):

    python tests/soak_stream.py [iterations]

Each iteration builds a random committee (ragged sizes), opens random
rounds, feeds shuffled vote batches with duplicates, wrong-message and
malformed signatures and out-of-range keys mixed in, and cross-checks every
round's exported bitmap + aggregate against the CPU oracle."""
import os
import random
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from harmony_amd import core  # noqa: E402
from oracle import capi, pyref as pr  # noqa: E402


def one_iteration(rng, it):
    n = rng.choice([17, 33, 64, 100, 256])
    R = rng.randint(1, 6)
    sks = [pr.fr_serialize(pr.synth_sk(1000 * it + i)) for i in range(n)]
    pks = core.batch_pk_from_sk(b"".join(sks), n)
    st = core.Stream(core.Committee(pks, n), R)
    payloads = [pr.construct_commit_payload(it, pr.synth_msg(10 * it + r), r)
                for r in range(R)]
    st.set_rounds(list(range(R)), b"".join(payloads), len(payloads[0]))
    expected = [set() for _ in range(R)]
    votes = []
    for r in range(R):
        for i in rng.sample(range(n), rng.randint(1, n)):
            votes.append((r, i, "ok"))
            if rng.random() < 0.15:
                votes.append((r, i, "dup"))
    for _ in range(rng.randint(0, 5)):
        votes.append((rng.randrange(R), rng.randrange(n), rng.choice(["bad", "mal"])))
    rng.shuffle(votes)
    for lo in range(0, len(votes), 37):
        chunk = votes[lo:lo + 37]
        sigs = b""
        for r, i, kind in chunk:
            if kind == "mal":
                sigs += b"\xff" * 96
            elif kind == "bad":
                sigs += capi.sign_hash(sks[i], b"x" * len(payloads[r]))
            else:
                sigs += capi.sign_hash(sks[i], payloads[r])
        res = st.process([v[1] for v in chunk], [v[0] for v in chunk], sigs)
        # group valid votes per (round, key): WITHIN a tick the dedup winner
        # is unordered (documented device semantics), so assert per group:
        # exactly one accept if the key was fresh, all duplicates otherwise.
        groups = {}
        for (r, i, kind), rc in zip(chunk, res):
            if kind in ("ok", "dup"):
                groups.setdefault((r, i), []).append(rc)
            else:
                assert rc <= 0, (it, r, i, kind, rc)
        for (r, i), rcs in groups.items():
            if i in expected[r]:
                assert all(rc == 2 for rc in rcs), (it, r, i, rcs)
            else:
                assert sorted(rcs) == [1] + [2] * (len(rcs) - 1), (it, r, i, rcs)
                expected[r].add(i)
    assert st.check(list(range(R))) == [True] * R
    oc = capi.Committee(pks, n)
    for r in range(R):
        bm, agg = st.get(r)
        want = bytearray((n + 7) // 8)
        for i in expected[r]:
            want[i >> 3] |= 1 << (i & 7)
        assert bm == bytes(want), (it, r)
        assert oc.agg_verify(bm, agg, payloads[r]) is True


def main():
    iters = int(sys.argv[1]) if len(sys.argv) > 1 else 10
    core.init()
    rng = random.Random(20260915)
    for it in range(iters):
        one_iteration(rng, it)
        print(f"iteration {it + 1}/{iters} ok", flush=True)
    print("soak_stream: PASS")


if __name__ == "__main__":
    main()
