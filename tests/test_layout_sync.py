"""Assert ops/csrc/l7_layout.h and store/l7_schema.py agree on column counts
and ordering-sensitive constants."""
import re
from pathlib import Path

from deepflow_amd.store import l7_schema as s

HDR = Path(__file__).resolve().parent.parent / "deepflow_amd/ops/csrc/l7_layout.h"


def _enum_count(text: str, terminator: str) -> int:
    # count enumerators before the *_N terminator in the enum containing it
    m = re.search(r"enum\s*\{([^}]*?)\b" + terminator + r"\b", text, re.S)
    assert m, terminator
    body = m.group(1)
    names = [ln.strip().split("=")[0].strip().rstrip(",")
             for ln in body.split(",") if ln.strip()]
    names = [n for n in names if n and not n.startswith("//")]
    return len(names)


def test_counts_match():
    text = HDR.read_text()
    # strip comments
    text = re.sub(r"//[^\n]*", "", text)
    assert _enum_count(text, "L7_U64_N") == s.N_U64
    assert _enum_count(text, "L7_U32_N") == s.N_U32
    assert _enum_count(text, "L7_U8_N") == s.N_U8
    assert _enum_count(text, "L7_STR_N") == s.N_STR
    assert _enum_count(text, "L7_DID_N") == s.N_DID
    assert _enum_count(text, "KG_VALS_N") == s.N_KG
    assert _enum_count(text, "DICT_DOM_N") == len(s.DICT_DOMAINS)
    m = re.search(r"#define L7_MAX_ATTRS (\d+)", text)
    assert int(m.group(1)) == s.MAX_ATTRS
