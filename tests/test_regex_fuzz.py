"""Differential regex fuzz: the host engine (regexutil fast paths + Glushkov
NFA, shared verbatim by the oracle and as a direct port by the HIP kernels)
against Python's `re` as an independent implementation, over the supported
pattern class (literals, classes, escape classes, quantifiers incl {m,n},
groups, alternation, top-level anchors) on ASCII subjects (where Python's
and Go's class semantics agree).

A 20k-pattern x 7-subject sweep ran clean during round 1; the committed test
is a bounded slice of the same generator."""

import ctypes
import random
import re as pyre

from victorialogs_amd import oracle_helpers

ALPHA = "abAB01 ._-"


def rand_atom(rng, depth):
    r = rng.random()
    if r < 0.35:
        return (pyre.escape(rng.choice(ALPHA))
                if rng.random() < 0.3 else rng.choice("abc01"))
    if r < 0.45:
        return "."
    if r < 0.60:
        neg = "^" if rng.random() < 0.3 else ""
        return "[%s%s]" % (neg, rng.choice(
            ["a-c", "0-9", "abc", "a-z", "xyz0-3"]))
    if r < 0.70:
        return rng.choice(["\\d", "\\w", "\\s", "\\D", "\\W", "\\S"])
    if depth <= 0:
        return rng.choice("ab01")
    if r < 0.85:
        return "(%s)" % rand_re(rng, depth - 1)
    return "(%s|%s)" % (rand_re(rng, depth - 1), rand_re(rng, depth - 1))


def rand_re(rng, depth):
    parts = []
    for _ in range(rng.randrange(1, 5)):
        a = rand_atom(rng, depth)
        q = rng.random()
        if q < 0.12:
            a += "*"
        elif q < 0.22:
            a += "+"
        elif q < 0.30:
            a += "?"
        elif q < 0.38:
            m = rng.randrange(0, 4)
            a += ("{%d}" % max(m, 1) if rng.random() < 0.5
                  else "{%d,%d}" % (m, m + rng.randrange(0, 3)))
        parts.append(a)
    return "".join(parts)


def test_regex_differential_fuzz():
    lib = oracle_helpers()
    lib.orc_regex_match.restype = ctypes.c_long
    rng = random.Random(7)
    checked = 0
    for _ in range(3000):
        pat = rand_re(rng, 2)
        if rng.random() < 0.25:
            pat = "^" + pat
        if rng.random() < 0.25:
            pat = pat + "$"
        try:
            cre = pyre.compile(pat)
        except pyre.error:
            continue
        pb = pat.encode()
        subjects = ["".join(rng.choice("abAB01 ._-cxyz3")
                            for _ in range(rng.randrange(0, 14)))
                    for _ in range(6)] + [""]
        rejected = False
        for s in subjects:
            sb = s.encode()
            r = lib.orc_regex_match(pb, len(pb), sb, len(sb))
            if r < 0:
                rejected = True  # outside the supported class: loud reject ok
                break
            want = 1 if cre.search(s) else 0
            assert r == want, f"pat={pat!r} s={s!r} ours={r} py={want}"
        if not rejected:
            checked += 1
    assert checked > 1500  # the generator mostly stays in the supported class
