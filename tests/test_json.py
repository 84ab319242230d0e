"""C++ JSON DOM edge cases (the parser behind znode payloads, config
files, and log lines)."""
import json

from binder_amd import require_native

n = require_native()


def rt(text):
    return n.json_roundtrip(text)


def test_basic_values():
    assert rt("null") == "null"
    assert rt("true") == "true"
    assert rt("-42") == "-42"
    assert rt('"hi"') == '"hi"'
    assert rt("[1,2,3]") == "[1,2,3]"
    assert rt('{"a":1}') == '{"a":1}'


def test_numbers():
    assert rt("0") == "0"
    assert rt("-0") == "0"
    assert rt("9007199254740993") == "9007199254740993"  # > 2^53
    assert json.loads(rt("1.5")) == 1.5
    assert json.loads(rt("1e3")) == 1000.0
    assert json.loads(rt("-2.5e-2")) == -0.025
    # int64 overflow falls back to double
    assert json.loads(rt("99999999999999999999999999")) > 1e25


def test_strings():
    assert rt(r'"\n\t\\\""') == r'"\n\t\\\""'
    assert json.loads(rt(r'"A"')) == "A"
    assert json.loads(rt(r'"😀"')) == "\U0001F600"  # surrogate
    assert json.loads(rt('"café"')) == "café"
    # raw control chars in input are rejected; escaped ones round-trip
    assert rt('"\x01"') is None
    assert "\\u0001" in rt('"\\u0001"')


def test_nesting_and_whitespace():
    assert rt('  { "a" : [ 1 , { "b" : [ ] } ] }  ') == \
        '{"a":[1,{"b":[]}]}'
    deep = "[" * 100 + "]" * 100
    assert rt(deep) is not None
    too_deep = "[" * 200 + "]" * 200
    assert rt(too_deep) is None  # depth cap, no stack overflow


def test_malformed_rejected():
    for bad in ["", "{", "[1,]", '{"a":}', '{"a" 1}', "tru", "nul",
                '"unterminated', "{}extra", "[1 2]", "+1", "'x'",
                '{"a":1,}', "NaN", "Infinity"]:
        assert rt(bad) is None, bad


def test_duplicate_keys_last_wins():
    # matches JSON.parse semantics the reference relies on
    assert rt('{"a":1,"a":2}') == '{"a":2}'


def test_python_json_agreement_fuzz():
    import random
    rng = random.Random(7)

    def gen(depth=0):
        choice = rng.randrange(7 if depth < 4 else 5)
        if choice == 0:
            return None
        if choice == 1:
            return rng.choice([True, False])
        if choice == 2:
            return rng.randint(-2**40, 2**40)
        if choice == 3:
            return round(rng.uniform(-1e6, 1e6), 6)
        if choice == 4:
            return "".join(chr(rng.randrange(32, 0x2FF))
                           for _ in range(rng.randrange(8)))
        if choice == 5:
            return [gen(depth + 1) for _ in range(rng.randrange(4))]
        return {f"k{i}": gen(depth + 1)
                for i in range(rng.randrange(4))}

    for _ in range(300):
        obj = gen()
        text = json.dumps(obj)
        out = rt(text)
        assert out is not None, text
        assert json.loads(out) == obj, (text, out)
