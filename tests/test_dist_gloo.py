"""Multi-process CPU tests (gloo, world_size=2) of the distributed semantics
the GPU path uses:

- config 3 (per-shard replicas): rank-sharded batches, no data exchange —
  results identical to single-rank runs item by item.
- config 4 (one 65536-key committee partitioned across ranks): each rank
  computes the masked PARTIAL pubkey sum over its index range, the partials
  are all-gathered and added — must equal the full-committee masked sum.
  EC addition is associative/commutative, so the split is bit-exact.

Crypto here runs on the CPU oracle (no GPU on the test host); the GPU path
shares the exact same partition + exchange shape via RCCL.
"""
import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

torch = pytest.importorskip("torch")
import torch.distributed as dist  # noqa: E402
import torch.multiprocessing as mp  # noqa: E402

N = 64          # committee size for the test (shape-identical to 65536)
WORLD = 2


def _init(rank, world, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)


def _worker_config4(rank, world, port, q):
    try:
        _init(rank, world, port)
        from oracle import capi, pyref as pr
        sks = [pr.fr_serialize(pr.synth_sk(i)) for i in range(N)]
        pks = [capi.pk_from_sk(s) for s in sks]
        bm = bytearray(N // 8)
        for i in range(N):
            if (i * 7) % 3 != 0:
                bm[i >> 3] |= 1 << (i & 7)
        # rank's index range [lo, hi)
        per = N // world
        lo, hi = rank * per, (rank + 1) * per
        # partial masked sum over the rank's slice (local bitmap slice)
        local_pks = pks[lo:hi]
        local_bm = bytearray(per // 8)
        for i in range(lo, hi):
            if bm[i >> 3] & (1 << (i & 7)):
                j = i - lo
                local_bm[j >> 3] |= 1 << (j & 7)
        comm = capi.Committee(b"".join(local_pks), per)
        partial = comm.mask_aggregate(bytes(local_bm))
        # exchange: all_gather of 48B serialized partials (the RCCL analog)
        t = torch.tensor(list(partial), dtype=torch.uint8)
        outs = [torch.zeros(48, dtype=torch.uint8) for _ in range(world)]
        dist.all_gather(outs, t)
        # reduce: EC add of the partials (on-GPU add tree in the real path)
        acc = bytes(outs[0].tolist())
        for o in outs[1:]:
            acc = capi.g1_add(acc, bytes(o.tolist()))
        # reference: full-committee masked sum
        full = capi.Committee(b"".join(pks), N).mask_aggregate(bytes(bm))
        q.put((rank, acc == full))
        dist.destroy_process_group()
    except Exception as e:  # pragma: no cover
        q.put((rank, f"error: {e}"))


def _worker_config3(rank, world, port, q):
    try:
        _init(rank, world, port)
        from oracle import capi, pyref as pr
        n = 8
        sks = [pr.fr_serialize(pr.synth_sk(i)) for i in range(n)]
        pks = [capi.pk_from_sk(s) for s in sks]
        comm = capi.Committee(b"".join(pks), n)
        # rank-sharded batch: items 2*rank, 2*rank+1 — no exchange at all
        results = []
        for j in (2 * rank, 2 * rank + 1):
            msg = pr.construct_commit_payload(j, pr.synth_msg(j), j)
            signers = [i for i in range(n) if (i + j) % 3 != 0]
            bm = bytearray(1)
            for i in signers:
                bm[0] |= 1 << i
            sk_sum = sum(pr.synth_sk(i) for i in signers) % pr.R
            sig = capi.sign_hash(pr.fr_serialize(sk_sum), msg)
            results.append(comm.agg_verify(bytes(bm), sig, msg))
        # gather results to rank 0 and check totals
        t = torch.tensor([int(all(results))])
        dist.all_reduce(t)
        q.put((rank, int(t.item()) == world))
        dist.destroy_process_group()
    except Exception as e:  # pragma: no cover
        q.put((rank, f"error: {e}"))


def _run(worker, port):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=worker, args=(r, WORLD, port, q)) for r in range(WORLD)]
    for p in procs:
        p.start()
    outs = [q.get(timeout=300) for _ in range(WORLD)]
    for p in procs:
        p.join(timeout=60)
    for rank, ok in outs:
        assert ok is True, f"rank {rank}: {ok}"


def test_config4_partial_sum_exchange(oracle_lib):
    _run(_worker_config4, 29511)


def test_config3_rank_sharded_replicas(oracle_lib):
    _run(_worker_config3, 29513)


def _worker_exchange_guard(rank, world, port, q):
    try:
        _init(rank, world, port)
        import bench
        batch, share = 6, 3
        i0 = rank * share
        # deterministic fake partials: byte = f(rank, item, pos)
        partials = bytes((rank * 101 + j * 7 + p) % 256
                         for j in range(batch) for p in range(48))
        ext = bench.exchange_partials(partials, rank, world, batch, i0, share,
                                      dist, "gloo")
        # expected: the OTHER rank's partials for MY items, same formula
        other = (rank + 1) % world
        want = bytes((other * 101 + j * 7 + p) % 256
                     for j in range(i0, i0 + share) for p in range(48))
        q.put((rank, ext == want))
        dist.destroy_process_group()
    except Exception as e:  # pragma: no cover
        q.put((rank, f"error: {e}"))


def test_config4_exchange_backend_guard():
    """the nccl(RCCL) and gloo branches of the config-4
    exchange share one byte path (bench.exchange_partials) — this pins the
    gathered layout/slicing, so an 8-GPU RCCL run exercises already-tested
    byte logic (only the tensor device differs)."""
    _run(_worker_exchange_guard, 29517)
