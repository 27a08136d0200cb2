"""Assert-matrix build test — counterpart of reference test_assert.c
compiled three ways + tools/test_assert.py: the three
assert tiers must trip exactly when the build type says they should."""
import os
import subprocess

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

PROG = r"""
#include "cimba/config.hpp"
#include <cstdio>
#include <cstdlib>
int main(int argc, char**) {
    if (argc == 2) cmb_assert_debug(false);
    if (argc == 3) cmb_assert_release(false);
    if (argc == 4) cmb_assert_always(false);
    puts("survived");
    return 0;
}
"""


@pytest.mark.parametrize("defines,trips", [
    ([], {"debug": True, "release": True, "always": True}),
    (["-DNDEBUG"], {"debug": False, "release": True, "always": True}),
    (["-DNDEBUG", "-DNASSERT"],
     {"debug": False, "release": False, "always": True}),
])
def test_assert_tiers(tmp_path, defines, trips):
    src = tmp_path / "a.cpp"
    src.write_text(PROG)
    exe = str(tmp_path / "a.out")
    support = os.path.join(ROOT, "cimba_amd", "csrc", "host", "support.cpp")
    cmd = ["g++", "-std=c++17", "-I",
           os.path.join(ROOT, "cimba_amd", "csrc", "include"), str(src),
           support, "-o", exe] + defines
    r = subprocess.run(cmd, capture_output=True, text=True)
    assert r.returncode == 0, r.stderr
    for tier, extra_args in (("debug", ["x"]), ("release", ["x", "x"]),
                             ("always", ["x", "x", "x"])):
        out = subprocess.run([exe] + extra_args, capture_output=True,
                             text=True)
        tripped = out.returncode != 0
        assert tripped == trips[tier], (defines, tier, out.returncode)
        if not tripped:
            assert "survived" in out.stdout
