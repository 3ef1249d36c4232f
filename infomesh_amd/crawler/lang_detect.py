"""Lightweight language detection: Unicode script ranges + tiny
word-frequency tables for 9 languages.
Reference parity: infomesh/crawler/lang_detect.py (same approach)."""
from __future__ import annotations

import re

_SCRIPTS = (
    ("ja", (0x3040, 0x30FF)),     # kana
    ("ko", (0xAC00, 0xD7AF)),     # hangul
    ("zh", (0x4E00, 0x9FFF)),     # han (ja also uses; kana checked first)
    ("ru", (0x0400, 0x04FF)),     # cyrillic
)

_MARKERS: dict[str, frozenset[str]] = {
    "en": frozenset("the and is of to in that it for was with are this".split()),
    "de": frozenset("der die das und ist nicht ein eine mit von für auf".split()),
    "fr": frozenset("le la les et est une dans pour que qui sur pas".split()),
    "es": frozenset("el la los que y es en un una para con del".split()),
    "it": frozenset("il la che e di un una per con del sono non".split()),
    "pt": frozenset("o que e de um uma para com não os do mais".split()),
    "nl": frozenset("de het een en van is dat niet met voor zijn op".split()),
}

_WORD_RE = re.compile(r"[a-zà-ÿäöüßñç]+", re.I)


def detect_language(text: str) -> str:
    """Best-effort ISO-639-1 code; '' when unknown."""
    if not text:
        return ""
    sample = text[:4000]
    counts = {}
    total_cjk = 0
    for ch in sample:
        cp = ord(ch)
        for lang, (lo, hi) in _SCRIPTS:
            if lo <= cp <= hi:
                counts[lang] = counts.get(lang, 0) + 1
                total_cjk += 1
                break
    letters = sum(1 for c in sample if not c.isspace())
    if letters and total_cjk / letters > 0.15:
        # kana presence dominates han for Japanese text
        if counts.get("ja", 0) > 0.1 * counts.get("zh", 1):
            ja = counts.get("ja", 0)
            if ja > 0 and ja * 4 >= counts.get("zh", 0):
                return "ja" if ja >= counts.get("ko", 0) else "ko"
        return max(counts, key=counts.get)
    words = [w.lower() for w in _WORD_RE.findall(sample)]
    if not words:
        return ""
    best, best_hits = "", 0
    for lang, markers in _MARKERS.items():
        hits = sum(1 for w in words if w in markers)
        if hits > best_hits:
            best, best_hits = lang, hits
    if best_hits >= max(2, len(words) // 50):
        return best
    return ""
