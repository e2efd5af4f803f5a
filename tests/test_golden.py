"""Golden-vector pins: the only cross-implementation artifacts the reference
repo provides.  These pin Fr decode, the herumi BLS_SWAP_G
base point, G1 scalar mult and G1 (de)serialization."""
import json
import os

import pytest

HERE = os.path.dirname(os.path.abspath(__file__))


def _golden():
    with open(os.path.join(HERE, "golden", "sk_pk.json")) as f:
        return json.load(f)


def _genesis():
    with open(os.path.join(HERE, "golden", "genesis_pubkeys.json")) as f:
        return json.load(f)


def test_golden_sk_pk_pyref():
    from oracle import pyref as pr
    vec = _golden()
    assert len(vec) >= 20
    for e in vec:
        sk = pr.fr_deserialize(bytes.fromhex(e["sk"]))
        pk = pr.get_public_key(sk)
        assert pr.g1_serialize(pk).hex() == e["pk"], e["pk"]


def test_golden_sk_pk_c_oracle(oracle_lib):
    for e in _golden():
        assert oracle_lib.pk_from_sk(bytes.fromhex(e["sk"])).hex() == e["pk"]


def test_base_point_derivation():
    """herumi getBasePoint == [h1] * FTmap_G1(1) — x AND y (oracle/pyref.py)."""
    from oracle import pyref as pr
    bp = pr.g1_mul(pr.ft_map_g1(1), pr.H1)
    assert bp == pr.HERUMI_G1
    assert pr.g1_in_subgroup(bp)


def test_genesis_corpus_c_oracle(oracle_lib):
    """All genesis pubkeys must deserialize (x<p, on curve, in subgroup)."""
    pks = _genesis()
    assert len(pks) > 3000
    bad = [h for h in pks if not oracle_lib.g1_check(bytes.fromhex(h))]
    assert bad == []


def test_genesis_sample_pyref():
    from oracle import pyref as pr
    for h in _genesis()[::200]:
        pt = pr.g1_deserialize(bytes.fromhex(h))
        assert pr.g1_serialize(pt).hex() == h


def test_psi_and_cofactor_identities():
    """psi(G2)=[z]G2 (asserted at pyref import) and BP-fast == [h_eff]P."""
    from oracle import pyref as pr
    p = pr.ft_map_g2((987654321, 123456789))
    assert pr.g2_clear_cofactor_fast(p) == pr.g2_mul(p, pr.H_EFF)
    assert pr.g2_in_subgroup(pr.g2_clear_cofactor_fast(p))
    assert pr.g2_in_subgroup(pr.g2_clear_cofactor_full(p))
