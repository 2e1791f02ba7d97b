"""Multi-pattern scanning compiler: regex-subset / literal lists → one DFA.

This is the shared front-end for every string-scanning plugin in the GPU
chain (deny_filter, regex_filter prefilter, pii_filter, harmful_content
keyword scan — reference: plugins/deny_filter/deny.py, plugins/regex_filter/
search_replace.py, cpex-pii-filter, plugins/harmful_content_detector/). The
reference runs Python `re` per request; here patterns are compiled ONCE into
a dense byte-class DFA that both the CPU reference scanner (oracle for
tests) and the HIP scan kernel (`ops/csrc/scan.hip`) execute. The DFA is a
*search* automaton (start-state always live), so it reports every position
at which any pattern ends — Hyperscan-style prefilter semantics.

Supported pattern language (covers every pattern the builtin plugins ship):
  literals, '.', escapes \\d \\w \\s \\D \\W \\S, char classes [a-z0-9_]
  (with ranges + negation), quantifiers ? * + {m} {m,} {m,n} on one atom.
Case-insensitive mode folds ASCII. No groups/alternation inside one pattern
— supply alternatives as separate patterns (they share the DFA anyway).

Output tables:
  next:   uint16[n_states][n_classes]  transition table
  klass:  uint8[256]                   byte → class
  accept: uint32[n_states]             bitmask of pattern ids (<=32/bank)
  max_len: per-pattern max match length (0 = unbounded) for span recovery
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, FrozenSet, List, Optional, Sequence, Tuple

import numpy as np

INF = -1  # unbounded repeat


# ---------------------------------------------------------------------------
# Pattern parsing → list of (byteset, min, max) atoms
# ---------------------------------------------------------------------------

_DIGITS = frozenset(range(0x30, 0x3A))
_WORD = frozenset(list(range(0x30, 0x3A)) + list(range(0x41, 0x5B)) + list(range(0x61, 0x7B)) + [0x5F])
_SPACE = frozenset([0x20, 0x09, 0x0A, 0x0D, 0x0B, 0x0C])
_ALL = frozenset(range(256))


def _fold(s: FrozenSet[int], ci: bool) -> FrozenSet[int]:
    if not ci:
        return s
    out = set(s)
    for b in s:
        if 0x41 <= b <= 0x5A:
            out.add(b + 0x20)
        elif 0x61 <= b <= 0x7A:
            out.add(b - 0x20)
    return frozenset(out)


def _parse_class(pat: str, i: int) -> Tuple[FrozenSet[int], int]:
    """Parse '[...]' starting at pat[i] == '['; returns (byteset, next_index)."""
    i += 1
    neg = False
    if i < len(pat) and pat[i] == "^":
        neg = True
        i += 1
    items: set = set()
    first = True
    while i < len(pat) and (pat[i] != "]" or first):
        first = False
        c = pat[i]
        if c == "\\" and i + 1 < len(pat):
            esc = pat[i + 1]
            base = {"d": _DIGITS, "w": _WORD, "s": _SPACE}.get(esc)
            if base is not None:
                items |= base
                i += 2
                continue
            c = esc
            i += 1
        if i + 2 < len(pat) and pat[i + 1] == "-" and pat[i + 2] != "]":
            lo, hi = ord(c), ord(pat[i + 2])
            items |= set(range(lo, hi + 1))
            i += 3
        else:
            items.add(ord(c))
            i += 1
    if i >= len(pat):
        raise ValueError(f"unterminated class in {pat!r}")
    i += 1  # skip ']'
    byteset = frozenset(items)
    if neg:
        byteset = _ALL - byteset
    return byteset, i


def _parse_quant(pat: str, i: int) -> Tuple[int, int, int]:
    """Parse optional quantifier at pat[i]; returns (min, max, next_index)."""
    if i >= len(pat):
        return 1, 1, i
    c = pat[i]
    if c == "?":
        return 0, 1, i + 1
    if c == "*":
        return 0, INF, i + 1
    if c == "+":
        return 1, INF, i + 1
    if c == "{":
        j = pat.index("}", i)
        body = pat[i + 1:j]
        if "," in body:
            lo_s, hi_s = body.split(",", 1)
            lo = int(lo_s)
            hi = INF if hi_s.strip() == "" else int(hi_s)
        else:
            lo = hi = int(body)
        return lo, hi, j + 1
    return 1, 1, i


def parse_pattern(pat: str, case_insensitive: bool = False) -> List[Tuple[FrozenSet[int], int, int]]:
    """Pattern → [(byteset, min, max)] atom list."""
    atoms: List[Tuple[FrozenSet[int], int, int]] = []
    i = 0
    while i < len(pat):
        c = pat[i]
        if c == "[":
            byteset, i = _parse_class(pat, i)
        elif c == "\\" and i + 1 < len(pat):
            esc = pat[i + 1]
            byteset = {
                "d": _DIGITS, "w": _WORD, "s": _SPACE,
                "D": _ALL - _DIGITS, "W": _ALL - _WORD, "S": _ALL - _SPACE,
            }.get(esc)
            if byteset is None:
                byteset = frozenset([ord(esc)])
            i += 2
        elif c == ".":
            byteset = _ALL - frozenset([0x0A])
            i += 1
        else:
            byteset = frozenset([ord(c)])
            i += 1
        lo, hi, i = _parse_quant(pat, i)
        atoms.append((_fold(byteset, case_insensitive), lo, hi))
    return atoms


# ---------------------------------------------------------------------------
# NFA (position automaton) → DFA subset construction
# ---------------------------------------------------------------------------

@dataclass
class ScanTables:
    next: np.ndarray        # uint16 [n_states, n_classes]
    klass: np.ndarray       # uint8 [256]
    accept: np.ndarray      # uint32 [n_states]
    n_states: int
    n_classes: int
    patterns: List[str]
    max_len: List[int]      # 0 = unbounded

    def nbytes(self) -> int:
        return self.next.nbytes + self.klass.nbytes + self.accept.nbytes


def compile_patterns(patterns: Sequence[str], case_insensitive: bool = False,
                     max_states: int = 20000) -> ScanTables:
    if len(patterns) == 0:
        raise ValueError("no patterns")
    if len(patterns) > 32:
        raise ValueError("max 32 patterns per DFA bank (use multiple banks)")

    # Expand each pattern's atoms into NFA states. NFA state = (pat_id, atom_idx,
    # rep_count) flattened: bounded repeats {m,n} expand to n copies (or m copies +
    # self-loop atom for unbounded).
    # nfa_trans: state -> list of (byteset, next_state); accept on reaching end.
    nfa_trans: List[List[Tuple[FrozenSet[int], int]]] = [[]]  # state 0 = start
    nfa_accept: Dict[int, int] = {}  # state -> pattern bitmask
    start_edges: List[Tuple[FrozenSet[int], int]] = nfa_trans[0]
    max_lens: List[int] = []

    for pid, pat in enumerate(patterns):
        atoms = parse_pattern(pat, case_insensitive)
        # compute max match length
        mlen = 0
        for (_bs, lo, hi) in atoms:
            if hi == INF:
                mlen = 0
                break
            mlen += hi
        max_lens.append(mlen)

        # Build a linear chain with optional skips.
        # chain entry points: list of state ids whose NEXT consumed byte begins here.
        # We construct states lazily: each consumed byte = an edge to a fresh state.
        # frontier = set of states from which the next atom's first byte departs.
        frontier = [0]

        def add_edge(srcs: List[int], byteset: FrozenSet[int]) -> int:
            dst = len(nfa_trans)
            nfa_trans.append([])
            for s in srcs:
                nfa_trans[s].append((byteset, dst))
            return dst

        for (byteset, lo, hi) in atoms:
            if hi == INF:
                # consume lo required, then self-loop
                for _ in range(max(lo, 1) if lo > 0 else 0):
                    dst = add_edge(frontier, byteset)
                    frontier = [dst]
                if lo == 0:
                    # zero-or-more: loop state reachable without consuming
                    dst = add_edge(frontier, byteset)
                    nfa_trans[dst].append((byteset, dst))
                    frontier = frontier + [dst]
                else:
                    nfa_trans[frontier[0]].append((byteset, frontier[0]))
            else:
                new_frontier: List[int] = []
                cur = frontier
                for k in range(hi):
                    dst = add_edge(cur, byteset)
                    if k + 1 >= lo:
                        new_frontier.append(dst)
                    cur = [dst]
                if lo == 0:
                    new_frontier = frontier + new_frontier
                frontier = new_frontier
        for s in frontier:
            nfa_accept[s] = nfa_accept.get(s, 0) | (1 << pid)

    # Byte equivalence classes: bytes with identical NFA-edge membership share a class.
    sig: Dict[int, list] = {b: [] for b in range(256)}
    for sidx, edges in enumerate(nfa_trans):
        for eidx, (byteset, dst) in enumerate(edges):
            for b in byteset:
                sig[b].append((sidx, eidx))
    klass_of: Dict[tuple, int] = {}
    klass = np.zeros(256, dtype=np.uint8)
    for b in range(256):
        key = tuple(sig[b])
        if key not in klass_of:
            klass_of[key] = len(klass_of)
        klass[b] = klass_of[key]
    n_classes = len(klass_of)
    if n_classes > 255:
        raise ValueError("too many byte classes")
    rep_byte = {}
    for b in range(256):
        rep_byte.setdefault(int(klass[b]), b)

    # Subset construction; start state always included (search semantics).
    start: FrozenSet[int] = frozenset([0])
    dfa_index: Dict[FrozenSet[int], int] = {start: 0}
    worklist = [start]
    rows: List[List[int]] = []
    accepts: List[int] = []

    def accept_mask(states: FrozenSet[int]) -> int:
        m = 0
        for s in states:
            m |= nfa_accept.get(s, 0)
        return m

    accepts.append(accept_mask(start))
    while worklist:
        cur = worklist.pop()
        idx = dfa_index[cur]
        while len(rows) <= idx:
            rows.append([0] * n_classes)
        for cls in range(n_classes):
            b = rep_byte[cls]
            nxt = {0}  # search: restart is always possible
            for s in cur:
                for (byteset, dst) in nfa_trans[s]:
                    if b in byteset:
                        nxt.add(dst)
            fz = frozenset(nxt)
            if fz not in dfa_index:
                if len(dfa_index) >= max_states:
                    raise ValueError("DFA state explosion")
                dfa_index[fz] = len(dfa_index)
                accepts.append(accept_mask(fz))
                worklist.append(fz)
            rows[idx][cls] = dfa_index[fz]
    n_states = len(dfa_index)
    next_tab = np.array(rows, dtype=np.uint16)
    return ScanTables(
        next=next_tab,
        klass=klass,
        accept=np.array(accepts, dtype=np.uint32),
        n_states=n_states,
        n_classes=n_classes,
        patterns=list(patterns),
        max_len=max_lens,
    )


def compile_literals(words: Sequence[str], case_insensitive: bool = True) -> ScanTables:
    """Literal word list → DFA (Aho-Corasick equivalent; deny_filter path)."""
    escaped = []
    for w in words:
        escaped.append("".join(ch if ch.isalnum() else "\\" + ch for ch in w))
    return compile_patterns(escaped, case_insensitive=case_insensitive)


# ---------------------------------------------------------------------------
# CPU reference scanner — the parity oracle for the HIP kernel
# ---------------------------------------------------------------------------

def scan_reference(tables: ScanTables, data: bytes) -> List[Tuple[int, int]]:
    """Scan one buffer; returns [(end_pos_exclusive, pattern_id)] for every match end."""
    nxt = tables.next
    klass = tables.klass
    accept = tables.accept
    state = 0
    out: List[Tuple[int, int]] = []
    for pos, b in enumerate(data):
        state = int(nxt[state, klass[b]])
        a = int(accept[state])
        while a:
            pid = (a & -a).bit_length() - 1
            out.append((pos + 1, pid))
            a &= a - 1
    return out


def match_mask_reference(tables: ScanTables, data: bytes) -> int:
    """Bitmask of pattern ids present anywhere in the buffer."""
    nxt = tables.next
    klass = tables.klass
    accept = tables.accept
    state = 0
    mask = 0
    for b in data:
        state = int(nxt[state, klass[b]])
        mask |= int(accept[state])
    return mask
