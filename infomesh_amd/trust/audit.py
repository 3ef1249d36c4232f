"""Random content audits: re-crawl a sampled URL and compare the
observed content hash against the stored attestation.

Reference parity: infomesh/trust/audit.py (~1 audit/hour/node, 3
independent auditors, auditor cross-validation, Merkle-proof audits).
Single-node adaptation: "auditors" are independent re-fetches (or
injected fetchers in tests); results feed TrustStore.record_audit per
source domain.
"""
from __future__ import annotations

import random
import time
from dataclasses import dataclass, field
from typing import Awaitable, Callable

from ..hashing import content_hash
from ..index.local_store import LocalStore, extract_domain
from .merkle import MerkleProof, MerkleTree
from .scoring import TrustStore

AUDITS_PER_HOUR = 1.0
N_AUDITORS = 3

FetchFn = Callable[[str], Awaitable[str | None]]  # url -> text | None


@dataclass
class AuditResult:
    url: str
    passed: bool
    votes: list[bool] = field(default_factory=list)
    reason: str = ""


class AuditScheduler:
    def __init__(self, store: LocalStore, trust: TrustStore,
                 fetch_fn: FetchFn, rate_per_hour: float = AUDITS_PER_HOUR,
                 auditors: int = N_AUDITORS, rng: random.Random | None = None,
                 detector=None):
        self.store = store
        self.trust = trust
        self.fetch_fn = fetch_fn
        self.rate = rate_per_hour
        self.auditors = auditors
        self.rng = rng or random.Random()
        self.last_audit = 0.0
        self.history: list[AuditResult] = []
        # trust/detector.MaliciousNodeDetector: failed audits feed the
        # threat ladder so repeat offenders escalate toward isolation.
        self.detector = detector

    def due(self, now: float | None = None) -> bool:
        now = now or time.time()
        return self.rate > 0 and (now - self.last_audit) >= 3600.0 / self.rate

    def pick_url(self) -> str | None:
        n = self.store.count()
        if n == 0:
            return None
        # sample a random doc id (ids may be sparse: retry a few times)
        for _ in range(10):
            row = self.store.conn.execute(
                "SELECT url FROM documents ORDER BY RANDOM() LIMIT 1"
            ).fetchone()
            if row:
                return row["url"]
        return None

    async def run_audit(self, url: str | None = None) -> AuditResult | None:
        self.last_audit = time.time()
        url = url or self.pick_url()
        if url is None:
            return None
        doc = self.store.get_document_by_url(url)
        if doc is None:
            return None
        votes: list[bool] = []
        for _ in range(self.auditors):
            text = await self.fetch_fn(url)
            if text is None:
                continue
            votes.append(content_hash(text) == doc.text_hash)
        if not votes:
            result = AuditResult(url, passed=True, votes=[],
                                 reason="unreachable — no verdict")
        else:
            # Content drift is normal on the live web: majority vote, and
            # an all-stale result only mildly penalizes (handled by EMA).
            passed = sum(votes) * 2 >= len(votes)
            result = AuditResult(url, passed=passed, votes=votes)
            domain = extract_domain(url)
            self.trust.record_audit(domain, passed)
            if self.detector is not None and not passed:
                threat = self.detector.record(domain, "audit_fail")
                if threat.isolate:
                    # detector verdict overrides the consecutive-
                    # failures ladder (CRITICAL always isolates)
                    self.trust.isolate(domain)
        self.history.append(result)
        return result


def merkle_audit(items: list[str], index: int
                 ) -> tuple[bytes, MerkleProof, bool]:
    """Build root + membership proof for item `index` and self-verify
    (reference: audit.py:360 Merkle-proof audits)."""
    tree = MerkleTree.from_items(items)
    proof = tree.prove(index)
    ok = MerkleTree.verify_proof(tree.root, proof, items[index])
    return tree.root, proof, ok


def cross_validate_auditors(votes_by_auditor: dict[str, bool]
                            ) -> tuple[bool, list[str]]:
    """Majority verdict + list of dissenting auditors (dishonest-auditor
    detection, reference: audit.py:474-506)."""
    if not votes_by_auditor:
        return True, []
    n_pass = sum(votes_by_auditor.values())
    verdict = n_pass * 2 >= len(votes_by_auditor)
    dissent = [a for a, v in votes_by_auditor.items() if v != verdict]
    return verdict, dissent
