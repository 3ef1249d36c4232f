"""Timezone-claim plausibility checks (anti off-peak-multiplier abuse).

Reference parity: infomesh/credits/timezone_verify.py (IP-offset
estimation vs claimed TZ). Offline adaptation: a node's *activity
histogram* (hour-of-day of its ledger entries) is compared against the
claimed timezone — a node claiming "off-peak 23:00-07:00 local" whose
LLM actions cluster in its claimed daytime is implausible.
"""
from __future__ import annotations

import time
from collections import Counter

from .ledger import CreditLedger, LLM_ACTIONS


def activity_histogram(ledger: CreditLedger,
                       claimed_utc_offset_h: float) -> Counter:
    """Hour-of-(claimed-local)-day histogram of LLM credit entries."""
    hist: Counter = Counter()
    for row in ledger.execute(
            "SELECT action, ts FROM credit_entries").fetchall():
        if row["action"] not in {a.value for a in LLM_ACTIONS}:
            continue
        local_h = int((row["ts"] / 3600.0 + claimed_utc_offset_h) % 24)
        hist[local_h] += 1
    return hist


def off_peak_fraction(hist: Counter, start_h: int = 23,
                      end_h: int = 7) -> float:
    total = sum(hist.values())
    if total == 0:
        return 1.0
    if start_h > end_h:
        in_window = sum(c for h, c in hist.items()
                        if h >= start_h or h < end_h)
    else:
        in_window = sum(c for h, c in hist.items()
                        if start_h <= h < end_h)
    return in_window / total


def verify_timezone_claim(ledger: CreditLedger,
                          claimed_utc_offset_h: float,
                          min_entries: int = 20,
                          min_off_peak_fraction: float = 0.6) -> dict:
    """Returns {"plausible", "off_peak_fraction", "n"}. A claim is
    implausible when most off-peak-multiplied work happened during the
    claimed local daytime."""
    hist = activity_histogram(ledger, claimed_utc_offset_h)
    n = sum(hist.values())
    frac = off_peak_fraction(hist)
    return {
        "plausible": n < min_entries or frac >= min_off_peak_fraction,
        "off_peak_fraction": round(frac, 3),
        "n": n,
    }
