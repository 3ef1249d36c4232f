"""Trust-layer tests: ed25519, keys, merkle, attestation, scoring,
audit, dmca, gdpr (reference parity: tests/test_merkle.py etc.)."""
from __future__ import annotations

import asyncio

from infomesh_amd.index.local_store import Document, LocalStore
from infomesh_amd.trust import ed25519
from infomesh_amd.trust.attestation import (attestation_batch_root,
                                            create_attestation,
                                            verify_attestation)
from infomesh_amd.trust.audit import (AuditScheduler, cross_validate_auditors,
                                      merkle_audit)
from infomesh_amd.trust.dmca import TakedownManager
from infomesh_amd.trust.gdpr import DeletionManager
from infomesh_amd.trust.keys import (KeyPair, ensure_keys, rotate_keys,
                                     verify_rotation)
from infomesh_amd.trust.merkle import MerkleTree
from infomesh_amd.trust.scoring import TrustStore, tier_of


def test_ed25519_rfc8032_vectors():
    seed = bytes.fromhex("9d61b19deffd5a60ba844af492ec2cc4"
                         "4449c5697b326919703bac031cae7f60")
    pub = ed25519.public_key(seed)
    assert pub.hex() == ("d75a980182b10ab7d54bfed3c964073a"
                         "0ee172f3daa62325af021a68f707511a")
    sig = ed25519.sign(seed, b"")
    assert sig.hex() == (
        "e5564300c360ac729086e2cc806e828a84877f1eb8e5d974d873e06522490155"
        "5fb8821590a33bacc61e39701cf9b46bd25bf5f0595bbe24655141438e7a100b")
    assert ed25519.verify(pub, b"", sig)
    assert not ed25519.verify(pub, b"tampered", sig)
    assert not ed25519.verify(pub, b"", sig[:-1] + b"\x00")


def test_keypair_roundtrip_and_persistence(tmp_path):
    kp = ensure_keys(tmp_path)
    kp2 = ensure_keys(tmp_path)
    assert kp.seed == kp2.seed and kp.node_id == kp2.node_id
    msg = b"attest this"
    assert KeyPair.verify(kp.public, msg, kp.sign(msg))


def test_key_rotation(tmp_path):
    old = ensure_keys(tmp_path)
    new, record = rotate_keys(tmp_path, old)
    assert verify_rotation(record)
    assert ensure_keys(tmp_path).public == new.public
    record["new_pub"] = "00" * 32
    assert not verify_rotation(record)


def test_merkle_tree_proofs():
    items = [f"doc-{i}" for i in range(7)]  # odd count exercises dup node
    tree = MerkleTree.from_items(items)
    for i in range(7):
        proof = tree.prove(i)
        assert MerkleTree.verify_proof(tree.root, proof, items[i])
        assert not MerkleTree.verify_proof(tree.root, proof, "wrong item")
    other = MerkleTree.from_items(items[:-1])
    assert other.root != tree.root


def test_merkle_empty_and_single():
    assert MerkleTree.from_items([]).root == b"\x00" * 32
    t = MerkleTree.from_items(["only"])
    assert MerkleTree.verify_proof(t.root, t.prove(0), "only")


def test_attestation_roundtrip():
    kp = KeyPair.generate()
    att = create_attestation(kp, "https://a.com/x", "rawhash", "texthash")
    assert verify_attestation(att)
    att.text_hash = "tampered"
    assert not verify_attestation(att)


def test_attestation_batch_root():
    kp = KeyPair.generate()
    atts = [create_attestation(kp, f"https://a.com/{i}", "r", "t")
            for i in range(3)]
    root = attestation_batch_root(atts)
    assert len(root) == 32 and root != b"\x00" * 32


def test_trust_scoring_tiers():
    ts = TrustStore()
    assert ts.tier("new.com") == "normal"
    for _ in range(6):
        ts.record_audit("good.com", True)
        ts.update_component("good.com", "contribution", 1.0)
        ts.update_component("good.com", "uptime", 1.0)
        ts.update_component("good.com", "summary_quality", 1.0)
    assert ts.score("good.com") > 0.8
    assert ts.tier("good.com") == "trusted"
    for _ in range(3):
        ts.record_audit("bad.com", False)
    assert ts.tier("bad.com") == "isolated"
    ts.record_audit("bad.com", True)  # recovery clears isolation
    assert ts.tier("bad.com") != "isolated"
    assert tier_of(0.1) == "untrusted"
    ts.close()


def test_audit_scheduler():
    store = LocalStore(":memory:")
    store.add_document(Document(url="https://a.com/1", text="stable text"))
    trust = TrustStore()

    async def good_fetch(url):
        return "stable text"

    async def bad_fetch(url):
        return "changed content entirely"

    async def run():
        sched = AuditScheduler(store, trust, good_fetch, auditors=3)
        assert sched.due()
        res = await sched.run_audit()
        assert res.passed and len(res.votes) == 3
        sched_bad = AuditScheduler(store, trust, bad_fetch, auditors=3)
        res2 = await sched_bad.run_audit()
        assert not res2.passed
    asyncio.run(run())
    store.close()
    trust.close()


def test_merkle_audit_and_cross_validation():
    root, proof, ok = merkle_audit(["a", "b", "c", "d"], 2)
    assert ok
    verdict, dissent = cross_validate_auditors(
        {"a1": True, "a2": True, "a3": False})
    assert verdict and dissent == ["a3"]


def test_dmca_takedown(seeded_store):
    kp = KeyPair.generate()
    tm = TakedownManager(seeded_store, kp)
    n = tm.file_notice("domain:docs.python.org", "copyright", "claimant@x")
    assert tm.verify_notice(n)
    assert tm.is_blocked("https://docs.python.org/3/tutorial/")
    removed = tm.apply_pending()
    assert removed == 2
    assert seeded_store.search("python tutorial") == []
    assert tm.overdue() == []
    tm.close()


def test_gdpr_deletion(seeded_store):
    kp = KeyPair.generate()
    dm = DeletionManager(seeded_store, kp)
    rec = dm.request_deletion("https://pytorch.org/docs/", reason="rtbf",
                              requester="user@example.com")
    assert "signature" in rec
    assert seeded_store.get_document_by_url("https://pytorch.org/docs/") is None
    assert dm.is_forgotten("https://pytorch.org/docs/")
    # re-adding then enforcing removes again
    seeded_store.add_document(Document(url="https://pytorch.org/docs/",
                                       text="new body"))
    assert dm.enforce() == 1
    assert len(dm.export_records()) == 1
    dm.close()


# ------------------------------------------------------- detector


def test_detector_escalation_and_isolation():
    from infomesh_amd.trust.detector import (
        MaliciousNodeDetector, ThreatLevel)
    t = [1000.0]
    d = MaliciousNodeDetector(now=lambda: t[0])
    assert d.assess("n1").level == ThreatLevel.NONE
    d.record("n1", "spam")
    assert d.assess("n1").level == ThreatLevel.NONE
    d.record("n1", "audit_fail")
    assert d.assess("n1").level >= ThreatLevel.MEDIUM
    th = d.record("n1", "invalid_signature", count=2)
    assert th.level >= ThreatLevel.HIGH and th.isolate
    # decay: far in the future the score drains back to NONE
    t[0] += 365 * 24 * 3600.0
    assert d.assess("n1").level == ThreatLevel.NONE
    assert d.threats() == []


def test_detector_unknown_kind_raises():
    from infomesh_amd.trust.detector import MaliciousNodeDetector
    import pytest as _pytest
    with _pytest.raises(ValueError):
        MaliciousNodeDetector().record("x", "nope")


def test_detector_threat_listing_sorted():
    from infomesh_amd.trust.detector import MaliciousNodeDetector
    d = MaliciousNodeDetector(now=lambda: 0.0)
    d.record("a", "spam")
    d.record("b", "invalid_proof", count=3)
    ths = d.threats()
    assert ths and ths[0].node_id == "b"


# ----------------------------------------------------- reputation


def test_reputation_grades_and_persistence(tmp_path):
    from infomesh_amd.trust.reputation import SummaryReputation
    p = tmp_path / "rep.json"
    r = SummaryReputation(path=p)
    assert r.get("n").grade == "?" and r.get("n").accept
    for _ in range(6):
        r.record("good", 0.95)
    for _ in range(8):
        r.record("bad", 0.05)
    assert r.get("good").grade == "A" and r.get("good").accept
    bad = r.get("bad")
    assert bad.grade == "F" and not bad.accept
    assert 0.0 <= r.summary_quality("bad") < 0.4
    lb = r.leaderboard()
    assert lb[0].node_id == "good"
    # reload from disk
    r2 = SummaryReputation(path=p)
    assert r2.get("good").samples == 6


def test_audit_failures_feed_detector(tmp_path):
    import asyncio
    from infomesh_amd.index.local_store import Document, LocalStore
    from infomesh_amd.trust.audit import AuditScheduler
    from infomesh_amd.trust.detector import MaliciousNodeDetector
    from infomesh_amd.trust.scoring import TrustStore
    store = LocalStore(tmp_path / "ls.db")
    store.add_document(Document(url="https://bad.example/x",
                                title="t", text="original body text"))
    trust = TrustStore(tmp_path / "trust.db")
    det = MaliciousNodeDetector()

    async def tampered_fetch(url):
        return "tampered content entirely different"

    sched = AuditScheduler(store, trust, tampered_fetch, detector=det)
    for _ in range(3):
        res = asyncio.run(sched.run_audit("https://bad.example/x"))
        assert res is not None and not res.passed
    threat = det.assess("bad.example")
    assert threat.events.get("audit_fail", 0) == 3
    assert threat.score > 0
    store.close()
    trust.close()
