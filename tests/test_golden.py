"""Fixed-seed stochastic golden regression — counterpart of reference
test/tools/test_stochastic.py (11 binaries run with seed 0x34f05c64d7ad598f,
stdout diffed against test/reference/*.txt).  Pins the RNG
streams, event ordering and model outputs bit-exactly; regenerate with
tests/regen_golden.py after an intentional change."""
import json
import os

import pytest

from .regen_golden import build

GOLDEN = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                      "golden.json")


@pytest.mark.skipif(not os.path.exists(GOLDEN), reason="golden.json missing")
def test_golden_outputs_stable():
    with open(GOLDEN) as f:
        want = json.load(f)
    got = json.loads(json.dumps(build()))  # normalize types
    for key in want:
        assert got[key] == want[key], f"golden mismatch in '{key}'"
