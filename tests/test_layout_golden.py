"""Layout/hash parity against golden vectors dumped from the REFERENCE'S
OWN headers (oracle/ref_dump.cpp -> tests/golden/hash_golden.csv).
Pins: ikey_t bitfield packing, TomasWang hash, iptr_t packing — for both
the oracle and the product library (DESIGN.md §4 pin 1)."""
import csv
import os

import wukong_amd as wk
from tests import oracle_util as ou

GOLD = os.path.join(os.path.dirname(__file__), "golden", "hash_golden.csv")


def _rows():
    with open(GOLD) as f:
        return list(csv.reader(f))


def test_golden_exists():
    rows = _rows()
    assert len(rows) > 70


def test_key_pack_and_hash_oracle():
    for r in _rows():
        if r[0] != "K":
            continue
        v, p, d, raw, h = (int(x) for x in r[1:])
        assert ou.key_pack(v, p, d) == raw
        assert ou.hash_u64(raw) == h


def test_key_pack_and_hash_product():
    for r in _rows():
        if r[0] != "K":
            continue
        v, p, d, raw, h = (int(x) for x in r[1:])
        assert wk.key_pack(v, p, d) == raw
        assert wk.hash_u64(raw) == h


def test_ptr_pack_product():
    for r in _rows():
        if r[0] != "P":
            continue
        s, o, t, raw = (int(x) for x in r[1:])
        assert wk.ptr_pack(s, o, t) == raw
