import os
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires a real MI355X (run via gpurun)"
    )


@pytest.fixture(scope="session")
def gen_part(tmp_path_factory):
    """A small vlogsgenerator-shaped part shared across tests."""
    from victorialogs_amd import generate_part

    d = str(tmp_path_factory.mktemp("parts") / "gen")
    generate_part(d, rows=30000, streams=2, rows_per_block=4096, msg_len=256,
                  seed=42)
    return d


@pytest.fixture(scope="session")
def typed_part(tmp_path_factory):
    """A custom part exercising every valueType, with edge-case values
    (filter_test.go fixture style)."""
    from victorialogs_amd import write_custom_part

    rows = 300
    ts = [1700000000000000000 + i * 1000000 for i in range(rows)]
    dict_vals = ["debug", "info", "warn", "error"]
    spec = {
        "blocks": [
            {
                "stream": 0,
                "timestamps": ts,
                "columns": [
                    {"name": "_msg", "values": [
                        f"log line {i} level={dict_vals[i % 4]} took {i % 50}ms"
                        for i in range(rows)]},
                    {"name": "lvl", "values": [dict_vals[i % 4] for i in range(rows)]},
                    {"name": "u8", "values": [str(i % 250) for i in range(rows)]},
                    {"name": "u16", "values": [str(300 + i * 7) for i in range(rows)]},
                    {"name": "u32", "values": [str(70000 + i * 1009) for i in range(rows)]},
                    {"name": "u64", "values": [str(5000000000 + i * 999999) for i in range(rows)]},
                    {"name": "i64", "values": [str((i - 150) * 37) for i in range(rows)]},
                    {"name": "f64", "values": [f"{(i - 150) / 8}" for i in range(rows)]},
                    {"name": "ip", "values": [f"10.{i % 256}.{(i * 3) % 256}.{(i * 7) % 256}"
                                              for i in range(rows)]},
                    {"name": "iso", "values": [
                        "2024-01-%02dT%02d:%02d:%02d.%03dZ"
                        % (1 + i % 28, i % 24, i % 60, (i * 3) % 60, i % 1000)
                        for i in range(rows)]},
                    {"name": "constcol", "values": ["fixed value 42"] * rows},
                    {"name": "mix", "values": [
                        ["12345", "1.5", "-7.25", "200ms", "1.5GiB", "abc def",
                         "3d2h", "100KB", "", "nan?"][i % 10] for i in range(rows)]},
                    {"name": "uni", "values": [
                        ("раз два три" if i % 3 == 0 else "foo bar") for i in range(rows)]},
                    # second columns for eq_field/le_field pairs
                    {"name": "lvl2", "values": [dict_vals[(i // 2) % 4] for i in range(rows)]},
                    {"name": "u8b", "values": [str((i + 1) % 250) for i in range(rows)]},
                    {"name": "i64b", "values": [str((i - 148) * 37) for i in range(rows)]},
                    {"name": "f64b", "values": [f"{(i - 149) / 8}" for i in range(rows)]},
                    {"name": "ipb", "values": [f"10.{(i + 1) % 256}.{(i * 3) % 256}.{(i * 7) % 256}"
                                               for i in range(rows)]},
                    {"name": "isob", "values": [
                        "2024-01-%02dT%02d:%02d:%02d.%03dZ"
                        % (1 + (i + 1) % 28, i % 24, i % 60, (i * 3) % 60, i % 1000)
                        for i in range(rows)]},
                    # parseMathNumber legs: scientific/hex/rfc3339/ipv4
                    {"name": "mix2", "values": [
                        ["1e5", "1.5e-3", "-2E2", "0x1F", "0o17", "0b101",
                         "2024-01-01T00:00:00Z", "2024-01-01T01:00:00+01:00",
                         "10.0.0.1", "1_000", "inf", "0x1.8p1"][i % 12]
                        for i in range(rows)]},
                    # stays ValueType string (a few unparseable entries)
                    {"name": "ipstr", "values": [
                        ("n/a" if i % 9 == 8 else f"172.16.{i % 256}.{(i * 5) % 256}")
                        for i in range(rows)]},
                ],
            },
            {
                # second block, different dict content and some empty values
                "stream": 1,
                "timestamps": [t + 10**12 for t in ts],
                "columns": [
                    {"name": "_msg", "values": [
                        f"other stream row {i}; ip=192.168.1.{i % 256}"
                        for i in range(rows)]},
                    {"name": "lvl", "values": [["fatal", "ERROR"][i % 2] for i in range(rows)]},
                    {"name": "u8", "values": [str(i % 7) for i in range(rows)]},
                ],
            },
        ]
    }
    d = str(tmp_path_factory.mktemp("parts") / "typed")
    write_custom_part(d, spec)
    return d


# A shared battery of filters used by both the CPU consistency tests and the
# GPU parity tests.  Every entry must compile on both paths.
FILTERS = [
    '{"type":"phrase","field":"_msg","phrase":"message"}',
    '{"type":"phrase","field":"_msg","phrase":"the stream 1 and"}',
    '{"type":"phrase","field":"_msg","phrase":"absent_token_zzz"}',
    '{"type":"phrase","field":"_msg","phrase":""}',
    '{"type":"phrase","field":"dict_0","phrase":"error"}',
    '{"type":"phrase","field":"missing_col","phrase":"x"}',
    '{"type":"phrase","field":"host","phrase":"host_0"}',
    '{"type":"exact","field":"host","value":"host_1"}',
    '{"type":"phrase","field":"u8_0","phrase":"17"}',
    '{"type":"phrase","field":"u16_0","phrase":"300"}',
    '{"type":"phrase","field":"u64_0","phrase":"123"}',
    '{"type":"phrase","field":"i64_0","phrase":"-55"}',
    '{"type":"regexp","field":"_msg","re":"stream (0|1) and"}',
    '{"type":"regexp","field":"_msg","re":"uuid=[0-9a-f]"}',
    '{"type":"regexp","field":"dict_0","re":"err(or)|info"}',
    '{"type":"regexp","field":"_msg","re":"ip=1.*uuid"}',
    '{"type":"time","min":1700000000000000000,"max":1700000004999000000}',
    '{"type":"time","min":0,"max":1}',
    '{"type":"range","field":"u8_0","min":10,"max":99.5}',
    '{"type":"range","field":"u64_0","min":0,"max":1e18}',
    '{"type":"range","field":"i64_0","min":-1e18,"max":0}',
    '{"type":"range","field":"float_0","min":1.25,"max":7.5}',
    '{"type":"and","filters":['
    '{"type":"phrase","field":"_msg","phrase":"message"},'
    '{"type":"phrase","field":"dict_0","phrase":"info"}]}',
    '{"type":"or","filters":['
    '{"type":"phrase","field":"dict_0","phrase":"error"},'
    '{"type":"phrase","field":"dict_0","phrase":"fatal"},'
    '{"type":"phrase","field":"dict_0","phrase":"INFO"}]}',
    '{"type":"not","filter":{"type":"phrase","field":"dict_0","phrase":"error"}}',
    '{"type":"and","filters":['
    '{"type":"time","min":1700000000000000000,"max":1700000019999000000},'
    '{"type":"not","filter":{"type":"regexp","field":"_msg","re":"stream 1"}},'
    '{"type":"or","filters":['
    '{"type":"phrase","field":"dict_0","phrase":"warn"},'
    '{"type":"phrase","field":"dict_1","phrase":"debug"}]}]}',
    # set/range filter family on the generated part
    '{"type":"in","field":"dict_0","values":["error","fatal","nosuch"]}',
    '{"type":"in","field":"u8_0","values":["17","42","9999"]}',
    '{"type":"in","field":"host","values":["host_0","host_3"]}',
    '{"type":"in","field":"_msg","values":[]}',
    '{"type":"contains_any","field":"_msg","values":["stream 0","uuid"]}',
    '{"type":"contains_all","field":"_msg","values":["message","stream","worker"]}',
    '{"type":"string_range","field":"dict_0","min":"e","max":"g"}',
    '{"type":"string_range","field":"u8_0","min":"1","max":"20"}',
    '{"type":"len_range","field":"_msg","min":100,"max":200}',
    '{"type":"len_range","field":"u16_0","min":3,"max":3}',
    '{"type":"day_range","start":0,"end":43200000000000}',
    '{"type":"week_range","start":0,"end":3}',
    '{"type":"value_type","field":"u8_0","value_type":"uint8"}',
    '{"type":"value_type","field":"dict_0","value_type":"dict"}',
]

TYPED_FILTERS = [
    '{"type":"phrase","field":"_msg","phrase":"level=error"}',
    '{"type":"phrase","field":"lvl","phrase":"warn"}',
    '{"type":"exact","field":"lvl","value":"ERROR"}',
    '{"type":"phrase","field":"u8","phrase":"13"}',
    '{"type":"phrase","field":"u16","phrase":"1000"}',
    '{"type":"phrase","field":"u32","phrase":"171512"}',
    '{"type":"phrase","field":"u64","phrase":"5099999901"}',
    '{"type":"phrase","field":"i64","phrase":"-37"}',
    '{"type":"phrase","field":"f64","phrase":"-18.625"}',
    '{"type":"phrase","field":"ip","phrase":"10.5.15.35"}',
    '{"type":"phrase","field":"ip","phrase":"10.5"}',
    '{"type":"phrase","field":"iso","phrase":"2024-01-03T02:02:06.002Z"}',
    '{"type":"phrase","field":"constcol","phrase":"fixed value 42"}',
    '{"type":"phrase","field":"constcol","phrase":"value"}',
    '{"type":"phrase","field":"uni","phrase":"два"}',
    '{"type":"exact","field":"u8","value":"3"}',
    '{"type":"exact","field":"f64","value":"0.125"}',
    '{"type":"exact","field":"ip","value":"10.9.27.63"}',
    '{"type":"range","field":"u8","min":100,"max":200}',
    '{"type":"range","field":"i64","min":-100.5,"max":100.5}',
    '{"type":"range","field":"f64","min":-1,"max":1}',
    '{"type":"range","field":"lvl","min":0,"max":1}',
    '{"type":"range","field":"ip","min":0,"max":4e9}',
    '{"type":"regexp","field":"lvl","re":"warn|error"}',
    '{"type":"regexp","field":"u8","re":"^?(1|2)"}'.replace("^?", ""),
    '{"type":"regexp","field":"ip","re":"192\\\\.168"}',
    '{"type":"regexp","field":"iso","re":"2024-01-0"}',
    '{"type":"regexp","field":"i64","re":"-3"}',
    # float64 slow paths (device Ryu formatting) + range on string column
    '{"type":"phrase","field":"f64","phrase":"18"}',
    '{"type":"phrase","field":"f64","phrase":"-18"}',
    '{"type":"phrase","field":"f64","phrase":"625"}',
    '{"type":"regexp","field":"f64","re":"18\\\\.625|0\\\\.5"}',
    '{"type":"regexp","field":"f64","re":"-1"}',
    '{"type":"range","field":"mix","min":0,"max":10000}',
    '{"type":"range","field":"mix","min":-10,"max":1.6e9}',
    # {m,n} repetition and top-level anchors (expanded into the NFA)
    '{"type":"regexp","field":"_msg","re":"to{1,2}k"}',
    '{"type":"regexp","field":"u8","re":"1{2}"}',
    '{"type":"regexp","field":"_msg","re":"^log line [0-9]{1,3} "}',
    '{"type":"regexp","field":"lvl","re":"^(warn|error)$"}',
    '{"type":"regexp","field":"u8","re":"^[0-9]{2}$"}',
    '{"type":"regexp","field":"mix","re":"ms$"}',
    '{"type":"regexp","field":"ip","re":"^10\\\\.2"}',
    # general regex class (Glushkov NFA on device)
    '{"type":"regexp","field":"_msg","re":"took \\\\d+ms"}',
    '{"type":"regexp","field":"_msg","re":"level=[a-z]+ took"}',
    '{"type":"regexp","field":"lvl","re":"(warn|err)o?r?"}',
    '{"type":"regexp","field":"mix","re":"\\\\d+KB|GiB"}',
    '{"type":"regexp","field":"uni","re":"р.з"}',
    '{"type":"regexp","field":"u8","re":"1\\\\d"}',
    '{"type":"regexp","field":"ip","re":"10\\\\.\\\\d+\\\\.1?5"}',
    # prefix / exact_prefix / sequence filters
    '{"type":"prefix","field":"_msg","prefix":"log li"}',
    '{"type":"prefix","field":"_msg","prefix":""}',
    '{"type":"prefix","field":"lvl","prefix":"warn"}',
    '{"type":"prefix","field":"u8","prefix":"1"}',
    '{"type":"prefix","field":"u16","prefix":"10"}',
    '{"type":"prefix","field":"i64","prefix":"-3"}',
    '{"type":"prefix","field":"i64","prefix":"-"}',
    '{"type":"prefix","field":"f64","prefix":"-18"}',
    '{"type":"prefix","field":"ip","prefix":"10.2"}',
    '{"type":"prefix","field":"iso","prefix":"2024-01-0"}',
    '{"type":"exact_prefix","field":"_msg","prefix":"log line 1"}',
    '{"type":"exact_prefix","field":"lvl","prefix":"ERR"}',
    '{"type":"exact_prefix","field":"u8","prefix":"13"}',
    '{"type":"exact_prefix","field":"i64","prefix":"-11"}',
    '{"type":"exact_prefix","field":"f64","prefix":"0."}',
    '{"type":"exact_prefix","field":"ip","prefix":"10.1"}',
    '{"type":"exact_prefix","field":"iso","prefix":"2024-01-1"}',
    '{"type":"sequence","field":"_msg","phrases":["log","level","took"]}',
    '{"type":"sequence","field":"_msg","phrases":["level=error","13ms"]}',
    '{"type":"sequence","field":"lvl","phrases":["warn"]}',
    '{"type":"sequence","field":"u8","phrases":["13"]}',
    '{"type":"sequence","field":"u8","phrases":["1","3"]}',
    '{"type":"sequence","field":"ip","phrases":["10","35"]}',
    '{"type":"sequence","field":"ip","phrases":["10.5.15.35"]}',
    '{"type":"sequence","field":"iso","phrases":["2024","002Z"]}',
    '{"type":"sequence","field":"f64","phrases":["18","625"]}',
    '{"type":"and","filters":['
    '{"type":"phrase","field":"lvl","phrase":"error"},'
    '{"type":"range","field":"u8","min":0,"max":50},'
    '{"type":"not","filter":{"type":"phrase","field":"_msg","phrase":"took 13ms"}}]}',
    '{"type":"or","filters":['
    '{"type":"phrase","field":"_msg","phrase":"level=warn"},'
    '{"type":"phrase","field":"_msg","phrase":"other stream row 7"}]}',
    # ---- set/range filter family (filter_in.go, filter_contains_any.go,
    #      filter_contains_all.go, filter_string_range.go, filter_ipv4_range.go,
    #      filter_len_range.go, filter_day_range.go, filter_week_range.go,
    #      filter_value_type.go, filter_stream_id.go) ----
    '{"type":"in","field":"lvl","values":["warn","fatal"]}',
    '{"type":"in","field":"lvl","values":["nosuch"]}',
    '{"type":"in","field":"u8","values":["13","200","5"]}',
    '{"type":"in","field":"u16","values":["300","1000","999999"]}',
    '{"type":"in","field":"u32","values":["171512","70000"]}',
    '{"type":"in","field":"u64","values":["5099999901"]}',
    '{"type":"in","field":"i64","values":["-37","0","37"]}',
    '{"type":"in","field":"f64","values":["-18.625","0.125","1e300"]}',
    '{"type":"in","field":"ip","values":["10.5.15.35","10.9.27.63"]}',
    '{"type":"in","field":"iso","values":["2024-01-03T02:02:06.002Z"]}',
    '{"type":"in","field":"_msg","values":["log line 1 level=info took 1ms"]}',
    '{"type":"in","field":"constcol","values":["fixed value 42"]}',
    '{"type":"in","field":"missing_col","values":["a",""]}',
    '{"type":"contains_any","field":"_msg","values":["level=error","13ms"]}',
    '{"type":"contains_any","field":"lvl","values":["warn","fatal"]}',
    '{"type":"contains_any","field":"u8","values":["13","5"]}',
    '{"type":"contains_any","field":"f64","values":["18.625","625"]}',
    '{"type":"contains_any","field":"ip","values":["10.5","192.168"]}',
    '{"type":"contains_any","field":"iso","values":["002Z"]}',
    '{"type":"contains_any","field":"i64","values":["-37"]}',
    '{"type":"contains_all","field":"_msg","values":["level=error","took"]}',
    '{"type":"contains_all","field":"_msg","values":["log","line","7"]}',
    '{"type":"contains_all","field":"u8","values":["13","13"]}',
    '{"type":"contains_all","field":"u8","values":["13","5"]}',
    '{"type":"contains_all","field":"f64","values":["18","625"]}',
    '{"type":"contains_all","field":"ip","values":["10","35"]}',
    '{"type":"contains_all","field":"iso","values":["2024","002Z"]}',
    '{"type":"string_range","field":"lvl","min":"e","max":"warn"}',
    '{"type":"string_range","field":"u8","min":"10","max":"20"}',
    '{"type":"string_range","field":"i64","min":"-5","max":"3"}',
    '{"type":"string_range","field":"f64","min":"-1","max":"1"}',
    '{"type":"string_range","field":"ip","min":"10.1","max":"10.3"}',
    '{"type":"string_range","field":"iso","min":"2024-01-1","max":"2024-01-2"}',
    '{"type":"string_range","field":"_msg","min":"log line 5","max":"log line 7"}',
    '{"type":"string_range","field":"lvl","min":"z","max":"a"}',
    '{"type":"ipv4_range","field":"ip","min":167777280,"max":167790000}',
    '{"type":"ipv4_range","field":"ip","min":168099840,"max":168165375}',
    '{"type":"ipv4_range","field":"ipstr","min":2886729728,"max":2886733823}',
    '{"type":"in","field":"ipstr","values":["172.16.3.15","n/a"]}',
    '{"type":"len_range","field":"ipstr","min":3,"max":3}',
    '{"type":"range","field":"mix2","min":0,"max":200000}',
    '{"type":"range","field":"mix2","min":-300,"max":40}',
    '{"type":"range","field":"mix2","min":1.7e18,"max":1.8e18}',
    '{"type":"range","field":"mix2","min":167772161,"max":167772161}',
    '{"type":"string_range","field":"ipstr","min":"172.16.1","max":"172.16.2"}',
    '{"type":"ipv4_range","field":"_msg","min":3232235776,"max":3232236031}',
    '{"type":"ipv4_range","field":"lvl","min":0,"max":4294967295}',
    '{"type":"len_range","field":"_msg","min":30,"max":33}',
    '{"type":"len_range","field":"lvl","min":4,"max":4}',
    '{"type":"len_range","field":"u8","min":1,"max":1}',
    '{"type":"len_range","field":"u16","min":4,"max":4}',
    '{"type":"len_range","field":"i64","min":1,"max":2}',
    '{"type":"len_range","field":"f64","min":1,"max":4}',
    '{"type":"len_range","field":"ip","min":10,"max":11}',
    '{"type":"len_range","field":"iso","min":24,"max":24}',
    '{"type":"len_range","field":"iso","min":1,"max":23}',
    '{"type":"len_range","field":"uni","min":7,"max":11}',
    '{"type":"day_range","start":80000000000000,"end":80000150000000}',
    '{"type":"day_range","start":0,"end":86399999999999}',
    '{"type":"day_range","start":79000000000000,"end":82000000000000,'
    '"offset":1000000000000}',
    '{"type":"week_range","start":2,"end":3}',
    '{"type":"week_range","start":3,"end":6}',
    '{"type":"week_range","start":3,"end":3,"offset":-90000000000000}',
    '{"type":"value_type","field":"lvl","value_type":"dict"}',
    '{"type":"value_type","field":"u8","value_type":"uint8"}',
    '{"type":"value_type","field":"constcol","value_type":"const"}',
    '{"type":"value_type","field":"f64","value_type":"float64"}',
    '{"type":"value_type","field":"missing_col","value_type":"string"}',
    '{"type":"stream_id","ids":[{"account":0,"project":0,"hi":"1","lo":"1"}]}',
    '{"type":"stream_id","ids":[{"account":0,"project":0,"hi":"0","lo":"1"},'
    '{"account":0,"project":0,"hi":"9","lo":"9"}]}',
    '{"type":"stream_id","ids":[]}',
    # ---- eq_field / le_field (filter_eq_field.go, filter_le_field.go) ----
    '{"type":"eq_field","field":"u8","other_field":"u8"}',
    '{"type":"eq_field","field":"lvl","other_field":"lvl2"}',
    '{"type":"eq_field","field":"u8","other_field":"u8b"}',
    '{"type":"eq_field","field":"u8","other_field":"u16"}',
    '{"type":"eq_field","field":"_msg","other_field":"lvl"}',
    '{"type":"eq_field","field":"constcol","other_field":"_msg"}',
    '{"type":"eq_field","field":"missing_col","other_field":"lvl"}',
    '{"type":"eq_field","field":"missing_col","other_field":"missing2"}',
    '{"type":"le_field","field":"u8","other_field":"u8b"}',
    '{"type":"le_field","field":"u8","other_field":"u8b","exclude_equal":true}',
    '{"type":"le_field","field":"u8","other_field":"u16"}',
    '{"type":"le_field","field":"i64","other_field":"i64b"}',
    '{"type":"le_field","field":"f64","other_field":"f64b"}',
    '{"type":"le_field","field":"f64","other_field":"u8"}',
    '{"type":"le_field","field":"lvl","other_field":"lvl2"}',
    '{"type":"le_field","field":"ip","other_field":"ipb"}',
    '{"type":"le_field","field":"iso","other_field":"isob"}',
    '{"type":"le_field","field":"_msg","other_field":"mix"}',
    '{"type":"le_field","field":"missing_col","other_field":"u8",'
    '"exclude_equal":true}',
    # ---- any-case filters (filter_any_case_phrase.go, filter_any_case_prefix.go) ----
    '{"type":"any_case_phrase","field":"lvl","phrase":"ERROR"}',
    '{"type":"any_case_phrase","field":"lvl","phrase":"error"}',
    '{"type":"any_case_phrase","field":"_msg","phrase":"LEVEL=WARN"}',
    '{"type":"any_case_phrase","field":"_msg","phrase":"Other STREAM"}',
    '{"type":"any_case_phrase","field":"uni","phrase":"ДВА"}',
    '{"type":"any_case_phrase","field":"uni","phrase":"FOO"}',
    '{"type":"any_case_phrase","field":"u8","phrase":"13"}',
    '{"type":"any_case_phrase","field":"f64","phrase":"-18.625"}',
    '{"type":"any_case_phrase","field":"ip","phrase":"10.5.15.35"}',
    '{"type":"any_case_phrase","field":"iso","phrase":"2024-01-03t02:02:06.002z"}',
    '{"type":"any_case_phrase","field":"constcol","phrase":"FIXED Value"}',
    '{"type":"any_case_phrase","field":"missing_col","phrase":""}',
    '{"type":"any_case_prefix","field":"lvl","prefix":"ERR"}',
    '{"type":"any_case_prefix","field":"lvl","prefix":"err"}',
    '{"type":"any_case_prefix","field":"_msg","prefix":"LOG Line"}',
    '{"type":"any_case_prefix","field":"uni","prefix":"РАЗ"}',
    '{"type":"any_case_prefix","field":"u8","prefix":"1"}',
    '{"type":"any_case_prefix","field":"iso","prefix":"2024-01-0"}',
    '{"type":"any_case_prefix","field":"mix","prefix":"1.5g"}',
    '{"type":"any_case_prefix","field":"missing_col","prefix":""}',
    '{"type":"and","filters":['
    '{"type":"in","field":"lvl","values":["error","fatal"]},'
    '{"type":"len_range","field":"_msg","min":1,"max":40},'
    '{"type":"week_range","start":2,"end":2}]}',
]
