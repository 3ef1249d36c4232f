"""Query NLP: stop words, synonym expansion, did-you-mean, filter parsing.

Reference parity: infomesh/search/nlp.py (stop words for 9 languages,
synonym query expansion, edit-distance did-you-mean, natural-language
filter parsing site:/lang:/dates → ParsedQuery, related-search tracker).
"""
from __future__ import annotations

import re
import time
from collections import OrderedDict
from dataclasses import dataclass, field

STOP_WORDS: dict[str, frozenset[str]] = {
    "en": frozenset("""a an and are as at be by for from has have he her his i in is it its
        of on or she that the their them they this to was we were what when where which who
        will with you your not no but if then so do does did can could should would about
        into over under after before between during above below again there here all any
        both each few more most other some such only own same than too very just""".split()),
    "de": frozenset("""der die das ein eine und oder aber nicht ist sind war waren ich du er
        sie es wir ihr mit von zu auf für in im am an als auch des dem den einer einem""".split()),
    "fr": frozenset("""le la les un une des et ou mais ne pas est sont était je tu il elle
        nous vous ils elles avec de du au aux pour dans sur par ce cette ces que qui""".split()),
    "es": frozenset("""el la los las un una unos unas y o pero no es son era yo tú él ella
        nosotros con de del al para en sobre por este esta estos estas que quien""".split()),
    "it": frozenset("""il lo la i gli le un uno una e o ma non è sono era io tu lui lei noi
        con di del al per in su da questo questa che chi""".split()),
    "pt": frozenset("""o a os as um uma uns umas e ou mas não é são era eu tu ele ela nós
        com de do da ao para em sobre por este esta que quem""".split()),
    "nl": frozenset("""de het een en of maar niet is zijn was ik jij hij zij wij met van
        naar op voor in aan als ook dit dat die""".split()),
    "ru": frozenset("""и в не на я он она оно мы вы они что это как но или же бы от до из
        у за по с к о для при так то все её его их""".split()),
    "ja": frozenset("""の に は を た が で て と し れ さ ある いる も する から な こと
        として い や など なっ ない この ため その あっ よう また もの""".split()),
}

_SYNONYMS: dict[str, list[str]] = {
    "fast": ["quick", "rapid"], "quick": ["fast"],
    "error": ["exception", "failure", "bug"], "bug": ["error", "defect"],
    "install": ["setup", "installation"], "setup": ["install"],
    "delete": ["remove", "erase"], "remove": ["delete"],
    "doc": ["documentation", "docs"], "docs": ["documentation"],
    "tutorial": ["guide", "howto"], "guide": ["tutorial"],
    "api": ["interface", "endpoint"],
    "config": ["configuration", "settings"], "settings": ["configuration"],
    "auth": ["authentication", "login"], "login": ["signin", "auth"],
    "example": ["sample", "demo"], "sample": ["example"],
    "performance": ["speed", "throughput"], "speed": ["performance"],
    "gpu": ["accelerator"], "ml": ["machine learning"],
    "db": ["database"], "database": ["db"],
    "async": ["asynchronous"], "sync": ["synchronous"],
}


def remove_stop_words(query: str, language: str = "en") -> str:
    sw = STOP_WORDS.get(language, STOP_WORDS["en"])
    kept = [t for t in query.split() if t.lower() not in sw]
    # Never empty the query entirely.
    return " ".join(kept) if kept else query


def expand_query(query: str, max_extra: int = 3) -> list[str]:
    """Return alternative query strings via the synonym table
    (used on sparse results — reference: search/query.py:136-156)."""
    terms = query.lower().split()
    out: list[str] = []
    for i, t in enumerate(terms):
        for syn in _SYNONYMS.get(t, []):
            alt = terms.copy()
            alt[i] = syn
            out.append(" ".join(alt))
            if len(out) >= max_extra:
                return out
    return out


# ------------------------------------------------------------ did-you-mean

def edit_distance(a: str, b: str, cap: int = 3) -> int:
    """Bounded Levenshtein distance."""
    if abs(len(a) - len(b)) > cap:
        return cap + 1
    prev = list(range(len(b) + 1))
    for i, ca in enumerate(a, 1):
        cur = [i]
        best = i
        for j, cb in enumerate(b, 1):
            c = min(prev[j] + 1, cur[j - 1] + 1, prev[j - 1] + (ca != cb))
            cur.append(c)
            best = min(best, c)
        if best > cap:
            return cap + 1
        prev = cur
    return prev[-1]


def did_you_mean(query: str, vocabulary: set[str] | dict[str, int],
                 max_distance: int = 2) -> str | None:
    """Suggest a corrected query when terms are near-misses of indexed
    vocabulary (reference: nlp.py:803-910). `vocabulary` may map
    term→frequency for tie-breaking."""
    freq = vocabulary if isinstance(vocabulary, dict) else {t: 1 for t in vocabulary}
    corrected: list[str] = []
    changed = False
    for term in query.split():
        tl = term.lower()
        if tl in freq or len(tl) < 3:
            corrected.append(term)
            continue
        best, best_d, best_f = None, max_distance + 1, 0
        for cand, f in freq.items():
            d = edit_distance(tl, cand, cap=max_distance)
            if d < best_d or (d == best_d and f > best_f):
                if d <= max_distance:
                    best, best_d, best_f = cand, d, f
        if best is not None:
            corrected.append(best)
            changed = True
        else:
            corrected.append(term)
    return " ".join(corrected) if changed else None


# ------------------------------------------------------- filter parsing

@dataclass
class ParsedQuery:
    text: str
    site: str | None = None
    language: str | None = None
    after: float | None = None
    before: float | None = None
    raw: str = ""


_FILTER_RE = re.compile(r"\b(site|lang|language|before|after):(\S+)", re.I)
_DATE_FMTS = ("%Y-%m-%d", "%Y/%m/%d", "%Y-%m", "%Y")


def _parse_date(text: str) -> float | None:
    for fmt in _DATE_FMTS:
        try:
            return time.mktime(time.strptime(text, fmt))
        except ValueError:
            continue
    return None


def parse_query_filters(query: str) -> ParsedQuery:
    """Extract site:/lang:/before:/after: filters from the query text
    (reference: nlp.py:911-971)."""
    pq = ParsedQuery(text=query, raw=query)
    def _sub(m: re.Match) -> str:
        key, val = m.group(1).lower(), m.group(2)
        if key == "site":
            pq.site = val.lower().lstrip("www.") if val.startswith("www.") else val.lower()
        elif key in ("lang", "language"):
            pq.language = val.lower()[:2]
        elif key == "before":
            pq.before = _parse_date(val)
        elif key == "after":
            pq.after = _parse_date(val)
        return ""
    pq.text = _FILTER_RE.sub(_sub, query).strip()
    pq.text = re.sub(r"\s{2,}", " ", pq.text)
    return pq


# ---------------------------------------------------- related searches

@dataclass
class RelatedSearchTracker:
    """Session-local co-occurrence tracker for `related searches`
    (reference: nlp.py:972+). Locked: the MCP HTTP transport records
    from many handler threads, and an unguarded OrderedDict iteration
    concurrent with a record() raises mid-request."""
    max_entries: int = 1000
    _recent: OrderedDict = field(default_factory=OrderedDict)
    _lock: "threading.Lock" = field(
        default_factory=lambda: __import__("threading").Lock())

    def record(self, query: str) -> None:
        q = query.strip().lower()
        if not q:
            return
        with self._lock:
            self._recent[q] = self._recent.pop(q, 0) + 1
            while len(self._recent) > self.max_entries:
                self._recent.popitem(last=False)

    def related(self, query: str, limit: int = 5) -> list[str]:
        terms = set(query.lower().split())
        if not terms:
            return []
        with self._lock:
            snapshot = list(self._recent.items())
        scored = []
        for past, cnt in snapshot:
            if past == query.lower():
                continue
            overlap = len(terms & set(past.split()))
            if overlap:
                scored.append((overlap * cnt, past))
        scored.sort(reverse=True)
        return [p for _, p in scored[:limit]]


def extract_keywords(text: str, top_n: int = 50, min_len: int = 2,
                     language: str = "en") -> list[tuple[str, int]]:
    """TF-based keyword extraction (reference: index/distributed.py:
    30-159, where it fed the DHT keyword->pointer publish; here the GPU
    shard indexes every term exhaustively, so this serves faceting,
    related-search seeding and summary key-fact hints).

    Returns (keyword, count) pairs, most frequent first, stop-worded
    and length-filtered, capped at top_n."""
    from collections import Counter
    sw = STOP_WORDS.get(language, STOP_WORDS["en"])
    words = re.findall(r"[\w'-]+", text.lower())
    counts = Counter(w for w in words
                     if len(w) >= min_len and w not in sw
                     and not w.isdigit())
    return counts.most_common(top_n)
