"""Credit subsystem tests (reference parity: tests/test_credits.py)."""
from __future__ import annotations

import random

from infomesh_amd.credits.farming import FarmingDetector
from infomesh_amd.credits.ledger import (Action, CreditLedger,
                                         OFF_PEAK_MULTIPLIER)
from infomesh_amd.credits.scheduling import EnergyAwareScheduler, is_off_peak
from infomesh_amd.credits.verification import CreditProofBuilder
from infomesh_amd.trust.keys import KeyPair


def _ledger(off_peak=False, kp=None):
    return CreditLedger(kp=kp, off_peak_fn=lambda ts: off_peak)


def test_action_weights():
    led = _ledger()
    led.record_action(Action.CRAWL, 10)        # +10
    led.record_action(Action.QUERY_SERVED, 4)  # +2
    led.record_action(Action.UPTIME, 2)        # +1
    assert abs(led.balance() - 13.0) < 1e-9
    led.close()


def test_off_peak_multiplier_llm_only():
    led = _ledger(off_peak=True)
    e = led.record_action(Action.LLM_SUMMARIZE, 1)
    assert e.multiplier == OFF_PEAK_MULTIPLIER
    e2 = led.record_action(Action.CRAWL, 1)
    assert e2.multiplier == 1.0
    led.close()


def test_tiers_and_search_cost():
    led = _ledger()
    assert led.tier() == 1 and led.search_cost() == 0.100
    led.record_action(Action.CRAWL, 150)
    assert led.tier() == 2 and led.search_cost() == 0.050
    led.record_action(Action.CRAWL, 1000)
    assert led.tier() == 3 and abs(led.search_cost() - 0.033) < 1e-9
    led.close()


def test_deduct_and_debt_mode():
    led = _ledger()
    cost = led.deduct_search_cost()
    assert cost == 0.100
    assert led.balance() < 0
    # inside grace window: not yet debt mode
    assert not led.in_debt_mode(now=led._debt_since or 0)
    # 73h later: debt mode doubles cost
    import time
    led.in_debt_mode()  # establish debt_since
    assert led.in_debt_mode(now=time.time() + 73 * 3600)
    led.close()


def test_hash_chain_verification(tmp_path):
    kp = KeyPair.generate()
    led = CreditLedger(tmp_path / "ledger.db", kp=kp,
                       off_peak_fn=lambda ts: False)
    for _ in range(5):
        led.record_action(Action.CRAWL, 1)
    assert led.verify_chain()
    # tamper
    led.execute("UPDATE credit_entries SET credits=999 WHERE id=3")
    led.commit()
    assert not led.verify_chain()
    led.close()


def test_credit_proof_roundtrip():
    kp = KeyPair.generate()
    led = _ledger(kp=kp)
    for i in range(10):
        led.record_action(Action.CRAWL, 1)
    pb = CreditProofBuilder(led, kp)
    proof = pb.build_proof(n_samples=3, rng=random.Random(0))
    assert CreditProofBuilder.verify_proof(proof)
    proof["samples"][0]["entry_hash"] = "tampered"
    assert not CreditProofBuilder.verify_proof(proof)
    led.close()


def test_off_peak_window():
    import time
    # 23:30 local is off-peak, 12:00 is not (defaults 23:00-07:00)
    t_noon = time.mktime((2024, 6, 1, 12, 0, 0, 0, 0, -1))
    t_night = time.mktime((2024, 6, 1, 23, 30, 0, 0, 0, -1))
    assert not is_off_peak(t_noon)
    assert is_off_peak(t_night)


def test_energy_scheduler_defers():
    s = EnergyAwareScheduler()
    import time
    t_noon = time.mktime((2024, 6, 1, 12, 0, 0, 0, 0, -1))
    t_night = time.mktime((2024, 6, 1, 23, 30, 0, 0, 0, -1))
    assert not s.submit("summarize", {"u": 1}, now=t_noon)
    assert s.submit("summarize", {"u": 2}, urgent=True, now=t_noon)
    assert s.drain(now=t_noon) == []
    drained = s.drain(now=t_night)
    assert drained == [("summarize", {"u": 1})]


def test_farming_detector():
    d = FarmingDetector()
    t0 = 1_000_000.0
    for i in range(150):
        d.record("llm_summarize", ts=t0 + i * 0.01)  # 150 in 1.5 s = burst
    assert d.on_probation(now=t0 + 1)
    assert d.multiplier(now=t0 + 1) == 0.0
    assert d.multiplier(now=t0 + 25 * 3600) == 1.0


def test_ledger_batched_accounting():
    """record_action_async coalesces events into one signed entry per
    action per flush; totals and the hash chain stay exact."""
    from infomesh_amd.credits.ledger import (ACTION_WEIGHTS, Action,
                                             CreditLedger)
    from infomesh_amd.trust.keys import KeyPair
    led = CreditLedger(":memory:", kp=KeyPair.generate())
    for _ in range(50):
        led.record_action_async(Action.QUERY_SERVED, 1.0)
    for _ in range(7):
        led.deduct_search_cost_async()
    n = led.flush_pending()
    assert n == 2   # one QUERY_SERVED entry + one SEARCH_SPEND entry
    st = led.stats()
    assert abs(st["by_action"]["query_served"]
               - 50 * ACTION_WEIGHTS[Action.QUERY_SERVED]) < 1e-9
    assert st["by_action"]["search_spend"] < 0
    assert led.verify_chain()
    led.close()
