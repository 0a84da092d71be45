#!/usr/bin/env python3
"""Ports the reference's per-filter test fixtures
(lib/logstorage/filter_*_test.go: the `columns := []column{...}` fixtures and
their `testFilterMatchForColumns(t, columns, f, col, []int{...})` expected row
sets) into committed golden fixtures (tests/golden/filter_fixtures.json).

Run from the repo root in the analysis container (needs /root/reference);
the generated JSON is committed so the test suite never reads the reference.
"""
import json
import os
import re
import sys

REF = "/root/reference/lib/logstorage"

FILTER_MAP = {
    "filterPhrase": ("phrase", {"fieldName": "field", "phrase": "phrase"}),
    "filterExact": ("exact", {"fieldName": "field", "value": "value"}),
    "filterPrefix": ("prefix", {"fieldName": "field", "prefix": "prefix"}),
    "filterExactPrefix": ("exact_prefix",
                          {"fieldName": "field", "prefix": "prefix"}),
    "filterSequence": ("sequence",
                       {"fieldName": "field", "phrases": "phrases"}),
    "filterAnyCasePhrase": ("any_case_phrase",
                            {"fieldName": "field", "phrase": "phrase"}),
    "filterAnyCasePrefix": ("any_case_prefix",
                            {"fieldName": "field", "prefix": "prefix"}),
    "filterIn": ("in", {"fieldName": "field", "values": "values"}),
    "filterContainsAny": ("contains_any",
                          {"fieldName": "field", "values": "values"}),
    "filterContainsAll": ("contains_all",
                          {"fieldName": "field", "values": "values"}),
    "filterRange": ("range", {"fieldName": "field", "minValue": "min",
                              "maxValue": "max"}),
    "filterStringRange": ("string_range",
                          {"fieldName": "field", "minValue": "min",
                           "maxValue": "max"}),
    "filterIPv4Range": ("ipv4_range",
                        {"fieldName": "field", "minValue": "min",
                         "maxValue": "max"}),
    "filterLenRange": ("len_range", {"fieldName": "field", "minLen": "min",
                                     "maxLen": "max"}),
    "filterRegexp": ("regexp", {"fieldName": "field", "re": "re"}),
    "filterValueType": ("value_type",
                        {"fieldName": "field", "valueType": "value_type"}),
    "filterEqField": ("eq_field", {"fieldName": "field",
                                   "otherFieldName": "other_field"}),
    "filterLeField": ("le_field",
                      {"fieldName": "field", "otherFieldName": "other_field",
                       "excludeEqualValues": "exclude_equal"}),
    # timestamp-harness filters (testFilterMatchForTimestamps)
    "filterTime": ("time", {"minTimestamp": "min", "maxTimestamp": "max"}),
    "filterDayRange": ("day_range", {"start": "start", "end": "end",
                                     "offset": "offset"}),
    "filterWeekRange": ("week_range", {"startDay": "start", "endDay": "end",
                                       "start": "start", "end": "end",
                                       "offset": "offset"}),
}


class P:
    def __init__(self, s):
        self.s = s
        self.i = 0

    def skip_ws(self):
        while self.i < len(self.s):
            c = self.s[self.i]
            if c in " \t\n\r,":
                self.i += 1
            elif self.s.startswith("//", self.i):
                j = self.s.find("\n", self.i)
                self.i = len(self.s) if j < 0 else j
            else:
                break

    def parse_go_string(self):
        s = self.s
        i = self.i
        if s[i] == "`":
            j = s.index("`", i + 1)
            self.i = j + 1
            return s[i + 1:j]
        assert s[i] == '"', s[i:i + 20]
        i += 1
        out = []
        while s[i] != '"':
            c = s[i]
            if c == "\\":
                e = s[i + 1]
                if e == "n": out.append("\n")
                elif e == "t": out.append("\t")
                elif e == "r": out.append("\r")
                elif e == '"': out.append('"')
                elif e == "\\": out.append("\\")
                elif e == "x":
                    out.append(chr(int(s[i + 2:i + 4], 16)))
                    i += 2
                elif e == "u":
                    out.append(chr(int(s[i + 2:i + 6], 16)))
                    i += 4
                else:
                    out.append(e)
                i += 2
            else:
                out.append(c)
                i += 1
        self.i = i + 1
        return "".join(out)

    def parse_string_list(self):
        # at "[]string{" ... or just "{"
        s = self.s
        j = s.index("{", self.i)
        self.i = j + 1
        vals = []
        while True:
            self.skip_ws()
            if self.s[self.i] == "}":
                self.i += 1
                return vals
            vals.append(self.parse_go_string())

    def parse_number(self):
        s = self.s
        m = re.match(r"-?(?:math\.Inf\((-?1)\)|inf|0[xX][0-9a-fA-F_]+|[0-9_]+(?:\.[0-9_]*)?(?:e-?[0-9]+)?|\(?1\s*<<\s*[0-9]+\)?(?:\s*-\s*1)?)",
                     s[self.i:])
        if not m:
            return None
        tok = m.group(0)
        self.i += len(tok)
        neg = tok.startswith("-")
        body = tok[1:] if neg else tok
        if "math.Inf" in body:
            v = float("inf") if "(1" in body.replace(" ", "") else float("-inf")
        elif body == "inf":
            v = float("inf")
        elif body[:2] in ("0x", "0X"):
            v = int(body.replace("_", ""), 16)
        elif "<<" in body:
            body2 = body.replace("(", "").replace(")", "").replace(" ", "")
            if body2.endswith("-1"):
                a, b = body2[:-2].split("<<")
                v = (int(a) << int(b)) - 1
            else:
                a, b = body2.split("<<")
                v = int(a) << int(b)
        else:
            body = body.replace("_", "")
            v = float(body) if ("." in body or "e" in body) else int(body)
        return -v if neg else v


def parse_columns(src, start):
    """Parses `[]column{ {name: ..., values: []string{...}}, ... }` at/after
    start; returns (columns_list, end_index)."""
    p = P(src)
    p.i = src.index("{", start) + 1  # outer []column{
    cols = []
    while True:
        p.skip_ws()
        if src[p.i] == "}":
            return cols, p.i + 1
        assert src[p.i] == "{", src[p.i:p.i + 30]
        p.i += 1
        name = None
        vals = None
        while True:
            p.skip_ws()
            if src[p.i] == "}":
                p.i += 1
                break
            if src.startswith("name:", p.i):
                p.i += 5
                p.skip_ws()
                name = p.parse_go_string()
            elif src.startswith("values:", p.i):
                p.i += 7
                p.skip_ws()
                vals = p.parse_string_list()
            else:
                raise ValueError("col field? " + src[p.i:p.i + 40])
        cols.append({"name": name, "values": vals})


def parse_filter(src, start):
    """Parses `&filterX{ field: value, ... }` (including nested and/or/not)
    ending before start's matching close; returns (json_tree, end) or
    (None, reason)."""
    m = re.match(r"&(filter\w+)\{", src[start:])
    if not m:
        return None, "not a filter literal"
    goname = m.group(1)
    if goname in ("filterAnd", "filterOr"):
        jtype = "and" if goname == "filterAnd" else "or"
        p = P(src)
        p.i = start + len(m.group(0))
        p.skip_ws()
        fm = re.match(r"filters:\s*\[\]filter\{", src[p.i:])
        if not fm:
            return None, "and/or shape"
        p.i += len(fm.group(0))
        subs = []
        while True:
            p.skip_ws()
            if src[p.i] == "}":
                p.i += 1
                break
            sub = parse_filter(src, p.i)
            if sub[0] is None:
                return None, "nested: " + sub[1]
            subs.append(sub[0])
            p.i = sub[1]
        p.skip_ws()
        if src[p.i] != "}":
            return None, "and/or close"
        return ({"type": jtype, "filters": subs}, p.i + 1)
    if goname == "filterNot":
        p = P(src)
        p.i = start + len(m.group(0))
        p.skip_ws()
        fm = re.match(r"f:\s*", src[p.i:])
        if not fm:
            return None, "not shape"
        p.i += len(fm.group(0))
        sub = parse_filter(src, p.i)
        if sub[0] is None:
            return None, "nested: " + sub[1]
        p.i = sub[1]
        p.skip_ws()
        if src[p.i] != "}":
            return None, "not close"
        return ({"type": "not", "filter": sub[0]}, p.i + 1)
    if goname not in FILTER_MAP:
        return None, f"unsupported {goname}"
    jtype, fmap = FILTER_MAP[goname]
    p = P(src)
    p.i = start + len(m.group(0))
    node = {"type": jtype}
    while True:
        p.skip_ws()
        if src[p.i] == "}":
            p.i += 1
            break
        fm = re.match(r"(\w+):", src[p.i:])
        if not fm:
            return None, "field? " + src[p.i:p.i + 30]
        fname = fm.group(1)
        p.i += len(fm.group(0))
        p.skip_ws()
        if fname not in fmap:
            return None, f"{goname}.{fname} unmapped"
        key = fmap[fname]
        if fname == "re":
            rm = re.match(r"mustCompileRegex\(", src[p.i:])
            if not rm:
                return None, "re literal?"
            p.i += len(rm.group(0))
            p.skip_ws()
            node[key] = p.parse_go_string()
            p.skip_ws()
            assert src[p.i] == ")"
            p.i += 1
        elif src[p.i] in "\"`":
            node[key] = p.parse_go_string()
        elif src.startswith("[]string{", p.i):
            node[key] = p.parse_string_list()
        elif src.startswith("true", p.i):
            node[key] = True
            p.i += 4
        elif src.startswith("false", p.i):
            node[key] = False
            p.i += 5
        else:
            v = p.parse_number()
            if v is None:
                return None, "value? " + src[p.i:p.i + 30]
            node[key] = v
    return (node, p.i)


def parse_int_list(src, start):
    p = P(src)
    p.i = src.index("{", start) + 1
    vals = []
    while True:
        p.skip_ws()
        if src[p.i] == "}":
            return vals, p.i + 1
        v = p.parse_number()
        if v is None:
            raise ValueError("int list: " + src[p.i:p.i + 30])
        vals.append(int(v))


GO_WEEKDAYS = {"time.Sunday": 0, "time.Monday": 1, "time.Tuesday": 2,
               "time.Wednesday": 3, "time.Thursday": 4, "time.Friday": 5,
               "time.Saturday": 6}


def port_ts_file(path):
    """Ports testFilterMatchForTimestamps-based fixtures (time / day_range /
    week_range): a timestamps list + filter + expected row indexes."""
    src = open(path).read()
    # resolve the week-range file's symbolic timestamps:
    #   sunday := time.Date(2024, 6, 9, 1, 0, 0, 0, time.UTC).UnixNano()
    import datetime
    for m in re.finditer(
            r"(\w+)\s*:=\s*time\.Date\((\d+),\s*(\d+),\s*(\d+),"
            r"\s*(\d+),\s*(\d+),\s*(\d+),\s*(\d+),\s*time\.UTC\)"
            r"\.UnixNano\(\)", src):
        y, mo, d, h, mi, sec, ns = (int(x) for x in m.groups()[1:])
        dt = datetime.datetime(y, mo, d, h, mi, sec,
                               tzinfo=datetime.timezone.utc)
        v = int(dt.timestamp()) * 10**9 + ns
        src = src.replace(m.group(0), "")
        src = re.sub(r"\b" + m.group(1) + r"\s*\+\s*(\d+)\*nsecsPerDay",
                     lambda g: str(v + int(g.group(1)) * 86400 * 10**9), src)
        src = re.sub(r"\b" + m.group(1) + r"\b(?!\w)", str(v), src)
    for name, val in GO_WEEKDAYS.items():
        src = src.replace(name, str(val))
    src = src.replace("offset: (12 * nsecsPerHour)", "offset: %d" % (12 * 3600 * 10**9))
    src = re.sub(r"offset:\s*\(?(-?\d+)\s*\*\s*nsecsPerHour\)?",
                 lambda g: "offset: %d" % (int(g.group(1)) * 3600 * 10**9), src)
    src = re.sub(r"offset:\s*\(?(-?\d+)\s*\*\s*nsecsPerDay\)?",
                 lambda g: "offset: %d" % (int(g.group(1)) * 86400 * 10**9), src)
    fixtures = []
    skipped = []
    ts_iter = [(m.start(), m) for m in
               re.finditer(r"timestamps\s*:?=\s*\[\]int64\{", src)]
    call_iter = list(re.finditer(
        r"testFilterMatchForTimestamps\(t,\s*timestamps,\s*(\w+),\s*(nil|\[\]int\{[^}]*\})\)",
        src))
    assigns = list(re.finditer(r"(\w+)\s*:?=\s*&(filter\w+)\{", src))
    for call in call_iter:
        pos = call.start()
        ts_def = None
        for tstart, tm in ts_iter:
            if tstart < pos:
                ts_def = tm
            else:
                break
        if ts_def is None:
            skipped.append("no timestamps")
            continue
        try:
            ts, _ = parse_int_list(src, ts_def.end() - 1)
        except ValueError:
            skipped.append("symbolic timestamps")
            continue
        var = call.group(1)
        adef = None
        for am in assigns:
            if am.start() < pos and am.group(1) == var:
                adef = am
            elif am.start() >= pos:
                break
        if adef is None:
            skipped.append(f"no assign for {var}")
            continue
        node = parse_filter(src, src.index("&", adef.start()))
        if node[0] is None:
            skipped.append(node[1])
            continue
        exp = call.group(2)
        rows = ([] if exp == "nil" else
                [int(x) for x in re.findall(r"-?\d+", exp)])
        fixtures.append({"timestamps": ts, "filter": node[0],
                         "expected": rows})
    return fixtures, skipped


def port_file(path):
    src = open(path).read()
    fixtures = []
    skipped = []
    # map of variable assignments: `xx := &filterX{` / `xx = &filterX{`
    columns = None
    col_iter = [(m.start(), m) for m in
                re.finditer(r"columns\s*:?=\s*\[\]column\{", src)]
    call_iter = list(re.finditer(
        r"testFilterMatchForColumns\(t,\s*columns,\s*(\w+),\s*\"((?:[^\"\\]|\\.)*)\",\s*(nil|\[\]int\{[^}]*\})\)",
        src))
    # assignments of filter vars
    assigns = list(re.finditer(r"(\w+)\s*:?=\s*&(filter\w+)\{", src))
    for call in call_iter:
        pos = call.start()
        # nearest preceding columns block
        cols_def = None
        for cstart, cm in col_iter:
            if cstart < pos:
                cols_def = cm
            else:
                break
        if cols_def is None:
            skipped.append("no columns")
            continue
        cols, _ = parse_columns(src, cols_def.end() - 1)
        var = call.group(1)
        # nearest preceding assignment of this var
        adef = None
        for am in assigns:
            if am.start() < pos and am.group(1) == var:
                adef = am
            elif am.start() >= pos:
                break
        if adef is None:
            skipped.append(f"no assign for {var}")
            continue
        node = parse_filter(src, src.index("&", adef.start()))
        if node[0] is None:
            skipped.append(node[1])
            continue
        # in/contains set their value list via `var.values.values = []string{...}`
        # after the struct literal (filter_in_test.go style)
        if (node[0].get("type") in ("in", "contains_any", "contains_all")
                and "values" not in node[0]):
            vm = None
            for am in re.finditer(
                    re.escape(var) + r"\.values\.values\s*=\s*\[\]string\{",
                    src):
                if adef.start() < am.start() < pos:
                    vm = am
            if vm is None:
                skipped.append(f"no values assign for {var}")
                continue
            p2 = P(src)
            p2.i = vm.end() - 1
            node[0]["values"] = p2.parse_string_list()
        exp = call.group(3)
        rows = ([] if exp == "nil" else
                [int(x) for x in re.findall(r"-?\d+", exp)])
        fixtures.append({"columns": cols, "filter": node[0],
                         "expected": rows})
    return fixtures, skipped


def main():
    out = {}
    ts_out = {}
    total = 0
    for fn in sorted(os.listdir(REF)):
        if not (fn.startswith("filter_") and fn.endswith("_test.go")):
            continue
        try:
            fx, sk = port_file(os.path.join(REF, fn))
        except Exception as e:
            print(f"{fn}: PARSE ERROR {e}")
            continue
        if fx:
            out[fn] = fx
            total += len(fx)
        tfx, tsk = port_ts_file(os.path.join(REF, fn))
        if tfx:
            ts_out[fn] = tfx
            total += len(tfx)
        from collections import Counter
        reasons = Counter(sk + tsk)
        print(f"{fn}: {len(fx)}+{len(tfx)} ported, {len(sk)+len(tsk)} skipped"
              + (f" {dict(reasons)}" if (sk or tsk) else ""))
    with open("tests/golden/filter_fixtures.json", "w") as f:
        json.dump(out, f, ensure_ascii=False, indent=0)
    with open("tests/golden/filter_ts_fixtures.json", "w") as f:
        json.dump(ts_out, f, ensure_ascii=False, indent=0)
    print("TOTAL", total)


if __name__ == "__main__":
    main()
