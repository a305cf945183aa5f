"""Kallsyms parser unit coverage (parca_agent_amd/kallsyms.py).

Reference resolves kernel frames via a kallsyms reader; these pin the
agent-side parser: text/weak-only filtering, kptr_restrict zeros,
bisect lookup semantics."""

from parca_agent_amd.kallsyms import Kallsyms


def _write(tmp_path, text):
    p = tmp_path / "kallsyms"
    p.write_text(text)
    return str(p)


def test_parse_and_lookup(tmp_path):
    k = Kallsyms(_write(tmp_path, (
        "ffffffff81000000 T _stext\n"
        "ffffffff81001000 t do_idle\n"
        "ffffffff81002000 W weak_fn\n"
        "ffffffff81003000 D some_data\n"      # data: filtered out
        "ffffffff81004000 r rodata_thing\n"   # rodata: filtered out
        "ffffffff81005000 T _etext\n")))
    assert len(k) == 4
    assert k.lookup(0xFFFFFFFF81000000) == "_stext"
    assert k.lookup(0xFFFFFFFF81001FFF) == "do_idle"   # mid-symbol
    assert k.lookup(0xFFFFFFFF81002000) == "weak_fn"   # W included
    assert k.lookup(0xFFFFFFFF81003500) == "weak_fn"   # data skipped
    assert k.lookup(0x1000) is None                    # below first


def test_kptr_restrict_zeros(tmp_path):
    k = Kallsyms(_write(tmp_path, (
        "0000000000000000 T hidden_a\n"
        "0000000000000000 t hidden_b\n")))
    assert len(k) == 0
    assert k.lookup(0) is None


def test_missing_file():
    k = Kallsyms("/nonexistent/kallsyms")
    assert len(k) == 0 and k.lookup(0xFFFF) is None


def test_malformed_lines(tmp_path):
    k = Kallsyms(_write(tmp_path, (
        "garbage\n"
        "\n"
        "ffffffff81001000 T good_sym\n"
        "notahex T bad\n")))
    # int() on "notahex" would raise; parser must survive or skip.
    assert k.lookup(0xFFFFFFFF81001000) == "good_sym"
