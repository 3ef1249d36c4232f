"""Tests for the remaining inventory components: types, pdf, credit
sync, timezone verify, github identity, port check, dx, peer handler."""
from __future__ import annotations

import zlib

from infomesh_amd.credits.github_identity import (detect_git_email,
                                                  ensure_owner_identity)
from infomesh_amd.credits.ledger import Action, CreditLedger
from infomesh_amd.credits.sync import (CreditSummary, CreditSyncStore,
                                       build_summary, owner_hash,
                                       verify_summary)
from infomesh_amd.credits.timezone_verify import verify_timezone_claim
from infomesh_amd.crawler.pdf import extract_pdf_text, looks_like_pdf
from infomesh_amd.summarizer.engine import (ExtractiveBackend,
                                            SummarizationEngine)
from infomesh_amd.summarizer.peer_handler import (PeerSummarizeHandler,
                                                  SummarizeRequest)
from infomesh_amd.trust.keys import KeyPair
from infomesh_amd.types import KeyPairLike, ShardLike
from infomesh_amd.utils.dx import generate_changelog, generate_tool_guide
from infomesh_amd.utils.port_check import (check_port_with_advice,
                                           detect_environment,
                                           find_free_port, port_available)


def test_types_protocols():
    assert isinstance(KeyPair.generate(), KeyPairLike)
    from infomesh_amd.index.gpu_index import CpuShard
    assert isinstance(CpuShard(), ShardLike)


def _tiny_pdf(text: str) -> bytes:
    content = f"BT /F1 12 Tf ({text}) Tj ET".encode()
    stream = zlib.compress(content)
    return (b"%PDF-1.4\n1 0 obj\n<< /Filter /FlateDecode /Length " +
            str(len(stream)).encode() + b" >>\nstream\n" + stream +
            b"\nendstream\nendobj\n%%EOF")


def test_pdf_extraction():
    pdf = _tiny_pdf("Hello PDF world of GPU kernels")
    assert looks_like_pdf(pdf)
    text = extract_pdf_text(pdf)
    assert "Hello PDF world" in text
    assert extract_pdf_text(b"not a pdf") == ""


def test_credit_sync_roundtrip():
    kp = KeyPair.generate()
    led = CreditLedger(kp=kp, off_peak_fn=lambda ts: False)
    led.record_action(Action.CRAWL, 10)
    s = build_summary(led, kp, "dev@example.com")
    assert verify_summary(s)
    tampered = CreditSummary.from_dict({**s.to_dict(), "balance": 999.0})
    assert not verify_summary(tampered)

    store = CreditSyncStore()
    assert store.ingest(s)
    assert not store.ingest(s)  # stale duplicate
    assert store.owner_total(owner_hash("dev@example.com"),
                             local_balance=5.0) == s.balance + 5.0
    led.close()
    store.close()


def test_credit_sync_export_import(tmp_path):
    kp = KeyPair.generate()
    led = CreditLedger(kp=kp, off_peak_fn=lambda ts: False)
    led.record_action(Action.CRAWL, 3)
    store = CreditSyncStore()
    store.ingest(build_summary(led, kp, "a@b.com"))
    assert store.export_dir(tmp_path / "sync") == 1
    store2 = CreditSyncStore()
    assert store2.import_dir(tmp_path / "sync") == 1
    led.close(); store.close(); store2.close()


def test_timezone_verify():
    led = CreditLedger(off_peak_fn=lambda ts: True)
    # all LLM entries at UTC midnight -> claimed offset 0 => hour 0 (off-peak)
    base = 86400.0 * 1000  # ts multiple of a day => hour 0 UTC
    for i in range(30):
        led.record_action(Action.LLM_SUMMARIZE, 1, ts=base + i * 86400)
    good = verify_timezone_claim(led, claimed_utc_offset_h=0)
    assert good["plausible"] and good["off_peak_fraction"] == 1.0
    bad = verify_timezone_claim(led, claimed_utc_offset_h=12)  # noon local
    assert not bad["plausible"]
    led.close()


def test_github_identity(tmp_path):
    # explicit email wins and persists
    assert ensure_owner_identity(tmp_path, "me@example.com") == "me@example.com"
    assert ensure_owner_identity(tmp_path) == "me@example.com"
    assert ensure_owner_identity(tmp_path / "x", "not-an-email") is None
    detect_git_email()  # must not raise


def test_port_check():
    p = find_free_port(38000)
    assert p and port_available(p)
    out = check_port_with_advice(p)
    assert out["available"] and "env" in out
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", p))
    s.listen(1)
    try:
        busy = check_port_with_advice(p)
        assert not busy["available"] and "advice" in busy
    finally:
        s.close()
    assert isinstance(detect_environment()["cloud"], str)


def test_dx_generators():
    guide = generate_tool_guide()
    assert "web_search" in guide and "Legacy aliases" in guide
    log = generate_changelog()
    assert log.startswith("# Changelog")


def test_peer_summarize_handler():
    eng = SummarizationEngine(backend=ExtractiveBackend())
    led = CreditLedger(off_peak_fn=lambda ts: False)
    h = PeerSummarizeHandler(eng, ledger=led, max_queue=2)
    res = h.handle(SummarizeRequest(
        text="The accelerator has many compute units. " * 10,
        title="T"))
    assert res is not None and h.stats["served"] == 1
    assert led.balance() > 0  # LLM_SUMMARIZE credit awarded
    assert h.handle(SummarizeRequest(text="")) is None
    assert h.handle(SummarizeRequest(text="x" * 200_000)) is None
    assert h.stats["rejected"] == 2
    led.close()
